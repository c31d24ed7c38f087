// Embedding-table gradient reduction: dE (12, 50) = Hot^T · dM without
// materialising anything — each workgroup streams a row chunk of dM,
// accumulates its local (12 x 50) table in LDS (atomic fp32 adds), and
// commits with 600 global atomics. Replaces both aten's scatter-add
// (9.1 ms/step) and the Tensile skinny-K GEMM (1.65 ms/step) — see
// profiles/train_breakdown_r01.

#include <cstdint>

#include "common.h"

namespace rk {

constexpr int NIDS = 12;
constexpr int ED = 50;

__global__ __launch_bounds__(256) void emb_grad_kernel(
    const bf16* __restrict__ dm,   // (N, ED) grads of the gathered rows
    const uint8_t* __restrict__ ids,  // (N)
    float* __restrict__ de,        // (NIDS, ED) pre-zeroed
    int64_t N) {
    __shared__ float acc[NIDS * ED];
    for (int e = threadIdx.x; e < NIDS * ED; e += 256) acc[e] = 0.0f;
    __syncthreads();

    // 256 threads = 5 rows x 50 cols (6 threads idle)
    const int e = threadIdx.x % ED;
    const int rsub = threadIdx.x / ED;
    const int64_t rows_per_wg = (N + gridDim.x - 1) / gridDim.x;
    const int64_t r0 = blockIdx.x * rows_per_wg;
    const int64_t r1 = min(r0 + rows_per_wg, N);
    if (rsub < 5) {
        for (int64_t r = r0 + rsub; r < r1; r += 5) {
            const float v = bf2f(dm[r * ED + e]);
            const int c = ids[r];
            atomicAdd(&acc[c * ED + e], v);
        }
    }
    __syncthreads();
    for (int i = threadIdx.x; i < NIDS * ED; i += 256)
        if (acc[i] != 0.0f) atomicAdd(&de[i], acc[i]);
}

void emb_grad(const void* dm, const uint8_t* ids, float* de, int64_t n,
              hipStream_t stream) {
    int blocks = 512;
    hipLaunchKernelGGL(emb_grad_kernel, dim3(blocks), dim3(256), 0, stream,
                       static_cast<const bf16*>(dm), ids, de, n);
}

}  // namespace rk
