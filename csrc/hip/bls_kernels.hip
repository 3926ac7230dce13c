// K1: batched BLS-over-BN254 signature verification on MI355X (gfx950).
//
// The auth-storm hot path (BASELINE.json config 2: 10k subscriber auths):
// the reference verifies one signature per connection on the CPU
// (cdn-proto/src/connection/auth/marshal.rs:66-72, broker.rs:266-273); here a
// batch of N pending auths is verified in one launch, one wavefront LANE per
// verification — the field tower (csrc/common/bn254*.h) is shared source
// with the host implementation, so device results are golden-tested
// bit-for-bit against host results.
//
// Register pressure: a full pairing needs several Fp12 temporaries
// (12 * 4 = 48 u64 each), far beyond the 512-VGPR file — the kernel spills
// to scratch by design; throughput comes from running 10k+ independent
// verifications across 256 CUs. A lane-cooperative Fp-parallel variant is a
// later optimization if profiling warrants it.

#include <hip/hip_runtime.h>
#include <stdint.h>

#include "../bls/bls.h"

using namespace bn254;

// __launch_bounds__(64, 1): one wave per SIMD unlocks the full 512-VGPR
// budget per lane — the pairing's working set (~110 u64 live values) spills
// ~5.4 KB/lane to scratch at the default budget, and scratch waits were
// ~74% of wave time (profiles/pmc_r01.txt). Occupancy is irrelevant here:
// the kernel is latency-bound per lane, not bandwidth- or wave-limited.
extern "C" __global__ void __launch_bounds__(64, 1)
k1_bls_verify(
    const uint8_t* __restrict__ vks,     // [N][128]
    const uint8_t* __restrict__ sigs,    // [N][64]
    uint8_t* __restrict__ msgs,          // flat namespaced messages, each with 1 spare byte
    const int64_t* __restrict__ moff,    // [N+1] offsets (end includes spare byte)
    int32_t N,
    int32_t* __restrict__ ok)            // [N] out: 1 valid, 0 invalid
{
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= N) return;

    bls::VerKey vk;
    if (!bls::verkey_deserialize(vks + (size_t)i * 128, vk)) { ok[i] = 0; return; }
    Fp sx, sy;
    if (!bls::sig_deserialize(sigs + (size_t)i * 64, sx, sy)) { ok[i] = 0; return; }
    uint8_t* scratch = msgs + moff[i];
    uint32_t msg_len = (uint32_t)(moff[i + 1] - moff[i] - 1);  // spare byte excluded
    ok[i] = bls::verify_core(vk, scratch, msg_len, sx, sy) ? 1 : 0;
}

// Device self-test: hash_to_g1 + sign-shaped scalar mul, for golden tests.
extern "C" __global__ void __launch_bounds__(64)
k1_hash_to_g1(
    uint8_t* __restrict__ msgs, const int64_t* __restrict__ moff, int32_t N,
    uint8_t* __restrict__ out)  // [N][64]
{
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= N) return;
    Fp hx, hy;
    uint8_t* scratch = msgs + moff[i];
    uint32_t len = (uint32_t)(moff[i + 1] - moff[i] - 1);
    if (bls::hash_to_g1_with_scratch(scratch, len, hx, hy)) {
        bls::sig_serialize(hx, hy, out + (size_t)i * 64);
    } else {
        for (int j = 0; j < 64; ++j) out[(size_t)i * 64 + j] = 0;
    }
}

extern "C" {

void launch_k1_bls_verify(const uint8_t* vks, const uint8_t* sigs, uint8_t* msgs,
                          const int64_t* moff, int32_t N, int32_t* ok, hipStream_t s) {
    int threads = 64;  // one wave per block: scratch-heavy kernel, keep blocks small
    int blocks = (N + threads - 1) / threads;
    hipLaunchKernelGGL(k1_bls_verify, dim3(blocks), dim3(threads), 0, s, vks, sigs, msgs,
                       moff, N, ok);
}

void launch_k1_hash_to_g1(uint8_t* msgs, const int64_t* moff, int32_t N, uint8_t* out,
                          hipStream_t s) {
    int threads = 64;
    int blocks = (N + threads - 1) / threads;
    hipLaunchKernelGGL(k1_hash_to_g1, dim3(blocks), dim3(threads), 0, s, msgs, moff, N, out);
}

}  // extern "C"
