#!/bin/bash
# ASan+UBSan lane for the native host core (the C++ rewrite
# replaces Rust's compile-time safety with sanitizer lanes).
set -e
cd "$(dirname "$0")/.."
mkdir -p build
g++ -O1 -g -std=c++17 -fsanitize=address,undefined -fno-omit-frame-pointer \
    -I csrc csrc/tests/test_native.cpp -o build/test_native_asan -lpthread
./build/test_native_asan
# TSan lane: the pump runs a live epoll thread — race-check it
g++ -O1 -g -std=c++17 -fsanitize=thread -fno-omit-frame-pointer \
    -I csrc csrc/tests/test_native.cpp -o build/test_native_tsan -lpthread
./build/test_native_tsan
echo "sanitizer lane OK"
