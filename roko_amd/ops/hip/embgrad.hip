// Embedding-table gradient reduction: dE (12, 50) = Hot^T · dM without
// materialising anything — each workgroup streams a row chunk of dM,
// accumulates its local (12 x 50) table in LDS (atomic fp32 adds), and
// commits with 600 global atomics. Replaces both aten's scatter-add
// (9.1 ms/step) and the Tensile skinny-K GEMM (1.65 ms/step) — see
// profiles/train_r01_kernel_stats.txt.
//
// The loop is 8-deep software-pipelined: all 8 global loads of an iteration
// issue before any LDS atomic consumes one, so the ~600-cycle HBM latency is
// paid once per iteration instead of once per element (the v1 kernel's
// load->atomic->load dependent chain measured 597 us; bandwidth math says
// ~30 us for the 230 MB of dM traffic).

#include <cstdint>

#include "common.h"

namespace rk {

constexpr int NIDS = 12;
constexpr int ED = 50;
constexpr int ROWS_PER_ITER = 5 * 8;  // 5 row lanes x 8-deep unroll

__global__ __launch_bounds__(256) void emb_grad_kernel(
    const bf16* __restrict__ dm,   // (N, ED) grads of the gathered rows
    const uint8_t* __restrict__ ids,  // (N)
    float* __restrict__ de,        // (NIDS, ED) pre-zeroed
    int64_t N) {
    __shared__ float acc_lds[NIDS * ED];
    for (int e = threadIdx.x; e < NIDS * ED; e += 256) acc_lds[e] = 0.0f;
    __syncthreads();

    // 256 threads = 5 rows x 50 cols (6 threads idle); each thread owns one
    // column of 8 rows per iteration (rows rsub, rsub+5, ..., rsub+35) and
    // accumulates a private per-class table in 12 REGISTERS via branchless
    // compare-select (3 VALU per class-element) — no atomics in the loop.
    const int e = threadIdx.x % ED;
    const int rsub = threadIdx.x / ED;
    float acc[NIDS];
#pragma unroll
    for (int c = 0; c < NIDS; ++c) acc[c] = 0.0f;
    const int64_t iters_total =
        (N + ROWS_PER_ITER - 1) / ROWS_PER_ITER;
    const int64_t iters_per_wg = (iters_total + gridDim.x - 1) / gridDim.x;
    const int64_t it0 = blockIdx.x * iters_per_wg;
    const int64_t it1 = min(it0 + iters_per_wg, iters_total);
    if (rsub < 5) {
        for (int64_t it = it0; it < it1; ++it) {
            const int64_t rbase = it * ROWS_PER_ITER + rsub;
            float v[8];
            int c[8];
#pragma unroll
            for (int q = 0; q < 8; ++q) {
                const int64_t r = rbase + (int64_t)q * 5;
                const bool ok = r < N;
                v[q] = ok ? bf2f(dm[r * ED + e]) : 0.0f;
                c[q] = ok ? ids[r] : 0;
            }
#pragma unroll
            for (int q = 0; q < 8; ++q)
#pragma unroll
                for (int cc = 0; cc < NIDS; ++cc)
                    acc[cc] += (c[q] == cc) ? v[q] : 0.0f;
        }
    }
#pragma unroll
    for (int cc = 0; cc < NIDS; ++cc)
        if (acc[cc] != 0.0f) lds_atomic_add(&acc_lds[cc * ED + e], acc[cc]);
    __syncthreads();
    for (int i = threadIdx.x; i < NIDS * ED; i += 256)
        if (acc_lds[i] != 0.0f) agent_atomic_add(&de[i], acc_lds[i]);
}

void emb_grad(const void* dm, const uint8_t* ids, float* de, int64_t n,
              hipStream_t stream) {
    int blocks = 1024;
    hipLaunchKernelGGL(emb_grad_kernel, dim3(blocks), dim3(256), 0, stream,
                       static_cast<const bf16*>(dm), ids, de, n);
}

}  // namespace rk
