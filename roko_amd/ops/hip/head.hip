// Per-position 5-class head: logits = h·W4^T + b4, optionally fused argmax
// (reference: rnn_model.py:59 fc4 + inference.py:116 argmax — SURVEY.md §2.4
// K5/K8). Skinny N=5 GEMV: one lane per (t, b) row, W4 staged in LDS,
// 16B-vectorised K loop on the VALU (MFMA has nothing to win at N=5).

#include <cstdint>

#include "common.h"

namespace rk {

constexpr int HH = 256;  // 2 * HIDDEN_SIZE
constexpr int NC = 5;    // classes

__global__ __launch_bounds__(256) void head_fwd_kernel(
    const bf16* __restrict__ hseq,  // (T, B, HH)
    const bf16* __restrict__ w4,    // (NC, HH)
    const float* __restrict__ b4,   // (NC)
    float* __restrict__ logits,     // (B, T, NC) or nullptr
    uint8_t* __restrict__ amax,     // (B, T) or nullptr
    int T, int B) {
    __shared__ bf16 w[NC][HH];
    __shared__ float bias[NC];
    const int tid = threadIdx.x;
    for (int e = tid; e < NC * HH; e += 256) (&w[0][0])[e] = w4[e];
    if (tid < NC) bias[tid] = b4[tid];
    __syncthreads();

    const int row = blockIdx.x * 256 + tid;  // row over T*B, layout (T, B)
    if (row >= T * B) return;
    const int t = row / B, b = row % B;
    const bf16* x = hseq + (size_t)row * HH;

    float acc[NC] = {0.f, 0.f, 0.f, 0.f, 0.f};
    for (int k = 0; k < HH; k += 8) {
        bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + k);
#pragma unroll
        for (int o = 0; o < NC; ++o) {
            bf16x8 wv = *reinterpret_cast<const bf16x8*>(&w[o][k]);
#pragma unroll
            for (int q = 0; q < 8; ++q)
                acc[o] += float(xv[q]) * float(wv[q]);
        }
    }
#pragma unroll
    for (int o = 0; o < NC; ++o) acc[o] += bias[o];

    if (logits) {
        float* dst = logits + ((size_t)b * T + t) * NC;
#pragma unroll
        for (int o = 0; o < NC; ++o) dst[o] = acc[o];
    }
    if (amax) {
        int best = 0;
        float bv = acc[0];
#pragma unroll
        for (int o = 1; o < NC; ++o)
            if (acc[o] > bv) { bv = acc[o]; best = o; }
        amax[(size_t)b * T + t] = uint8_t(best);
    }
}

void head_fwd(const void* hseq, const void* w4, const float* b4, float* logits,
              uint8_t* amax, int T, int B, hipStream_t stream) {
    int rows = T * B;
    hipLaunchKernelGGL(head_fwd_kernel, dim3((rows + 255) / 256), dim3(256), 0,
                       stream, static_cast<const bf16*>(hseq),
                       static_cast<const bf16*>(w4), b4, logits, amax, T, B);
}

}  // namespace rk
