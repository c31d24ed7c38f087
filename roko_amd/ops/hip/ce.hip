// Fused 5-class cross-entropy: forward loss AND input gradient in one pass
// (reference op: F.cross_entropy(logits.transpose(1,2), y), train.py:50 —
// SURVEY.md §2.4 K6). With 5 classes a lane owns a full row: log-softmax,
// NLL and (softmax - onehot)/N all stay in registers; the scalar loss is a
// two-level reduction (wave shuffle + one global atomic per wave).

#include <cstdint>

#include "common.h"

namespace rk {

constexpr int NCLS = 5;

__global__ __launch_bounds__(256) void ce_fwd_bwd_kernel(
    const float* __restrict__ logits,  // (N, NCLS) rows
    const int64_t* __restrict__ target,  // (N)
    float* __restrict__ dlogits,       // (N, NCLS) out: d(mean loss)/dlogits
    float* __restrict__ loss_sum,      // (1) accumulated sum of row losses
    int64_t N) {
    const int64_t row = (int64_t)blockIdx.x * 256 + threadIdx.x;
    float loss = 0.0f;
    if (row < N) {
        const float* x = logits + row * NCLS;
        float m = x[0];
#pragma unroll
        for (int c = 1; c < NCLS; ++c) m = fmaxf(m, x[c]);
        float e[NCLS], s = 0.0f;
#pragma unroll
        for (int c = 0; c < NCLS; ++c) {
            e[c] = __expf(x[c] - m);
            s += e[c];
        }
        const float inv_s = 1.0f / s;
        const int t = int(target[row]);
        loss = logf(s) - (x[t] - m);
        const float inv_n = 1.0f / float(N);
        float* d = dlogits + row * NCLS;
#pragma unroll
        for (int c = 0; c < NCLS; ++c)
            d[c] = (e[c] * inv_s - (c == t ? 1.0f : 0.0f)) * inv_n;
    }
    // wave-level sum, one atomic per wave
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) loss += __shfl_down(loss, off, 64);
    if ((threadIdx.x & 63) == 0) agent_atomic_add(loss_sum, loss);
}

void ce_fwd_bwd(const float* logits, const int64_t* target, float* dlogits,
                float* loss_sum, int64_t n, hipStream_t stream) {
    const int64_t blocks = (n + 255) / 256;
    hipLaunchKernelGGL(ce_fwd_bwd_kernel, dim3((uint32_t)blocks), dim3(256), 0,
                       stream, logits, target, dlogits, loss_sum, n);
}

}  // namespace rk
