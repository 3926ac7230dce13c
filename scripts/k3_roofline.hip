// Standalone microbench: K3's write pattern (1120B records scattered across
// 10k per-user rings) with NO source reads — isolates whether K3 is bound
// by its read side (L2 message reads + pair loads) or by DRAM write
// locality of the ring scatter itself.
#include <hip/hip_runtime.h>
#include <stdio.h>
#include <stdint.h>

typedef unsigned int v4u __attribute__((ext_vector_type(4)));

// write pattern identical to k3_fanout_flat_t: unit-per-lane flat index,
// NT stores, pairs grouped by user (user-major order)
__global__ void __launch_bounds__(256) k3_writeonly(
    int64_t n_pairs, int32_t units, int64_t ring_bytes, int32_t n_msgs,
    uint64_t rec, uint8_t* egress)
{
    const int64_t n_units = n_pairs * units;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    v4u v = {1u, 2u, 3u, 4u};
    for (int64_t f = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; f < n_units; f += stride) {
        const int64_t p = f / units;
        const int32_t unit = (int32_t)(f - p * units);
        const int64_t u = p / n_msgs;          // user-major like K2b's pair list
        const int64_t m = p - u * n_msgs;
        uint8_t* dst = egress + u * ring_bytes + m * rec + (size_t)unit * 16;
        __builtin_nontemporal_store(v, (v4u*)dst);
    }
}

// sequential variant: same byte count, contiguous addresses
__global__ void __launch_bounds__(256) seq_writeonly(int64_t n16, uint8_t* out)
{
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    v4u v = {1u, 2u, 3u, 4u};
    for (int64_t f = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; f < n16; f += stride)
        __builtin_nontemporal_store(v, ((v4u*)out) + f);
}

int main() {
    const int64_t n_users = 10000, n_msgs = 256;
    const uint64_t rec = 1120;       // ring_rec(1104) for the 1 KiB bench
    const int32_t units = rec / 16;
    const int64_t ring_bytes = 1 << 20;
    const int64_t n_pairs = n_users * n_msgs;
    uint8_t* egress;
    (void)hipMalloc(&egress, n_users * ring_bytes);
    const int64_t bytes = n_pairs * (int64_t)rec;
    hipEvent_t a, b; (void)hipEventCreate(&a); (void)hipEventCreate(&b);
    for (int grid : {16384, 8192, 32768}) {
        for (int i = 0; i < 3; i++)
            hipLaunchKernelGGL(k3_writeonly, dim3(grid), dim3(256), 0, 0,
                               n_pairs, units, ring_bytes, (int32_t)n_msgs, rec, egress);
        (void)hipEventRecord(a);
        for (int i = 0; i < 10; i++)
            hipLaunchKernelGGL(k3_writeonly, dim3(grid), dim3(256), 0, 0,
                               n_pairs, units, ring_bytes, (int32_t)n_msgs, rec, egress);
        (void)hipEventRecord(b); (void)hipEventSynchronize(b);
        float ms; (void)hipEventElapsedTime(&ms, a, b); ms /= 10;
        printf("k3-pattern grid=%d: %.0f us, %.2f TB/s\n", grid, ms * 1000,
               bytes / (ms / 1000.0) / 1e12);
    }
    const int64_t n16 = bytes / 16;
    for (int i = 0; i < 3; i++)
        hipLaunchKernelGGL(seq_writeonly, dim3(16384), dim3(256), 0, 0, n16, egress);
    (void)hipEventRecord(a);
    for (int i = 0; i < 10; i++)
        hipLaunchKernelGGL(seq_writeonly, dim3(16384), dim3(256), 0, 0, n16, egress);
    (void)hipEventRecord(b); (void)hipEventSynchronize(b);
    float ms; (void)hipEventElapsedTime(&ms, a, b); ms /= 10;
    printf("sequential:   %.0f us, %.2f TB/s\n", ms * 1000, bytes / (ms / 1000.0) / 1e12);
    return 0;
}
