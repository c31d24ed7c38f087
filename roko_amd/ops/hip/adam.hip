// Fused Adam over ONE flat parameter buffer (reference op: torch Adam over
// ~1.1 M params, train.py:39 — SURVEY.md §2.4 K7). The optimizer keeps all
// parameters, gradients and moments in single flat fp32 tensors (the model's
// param tensors are views), so the whole update is one elementwise kernel —
// no per-tensor launch storm, one RCCL all-reduce on the flat grad in DP.

#include <cstdint>

#include "common.h"

namespace rk {

__global__ __launch_bounds__(256) void adam_step_kernel(
    float* __restrict__ p, const float* __restrict__ g, float* __restrict__ m,
    float* __restrict__ v, int64_t n, float lr, float beta1, float beta2,
    float eps, float bc1, float bc2, const int* __restrict__ step_ptr) {
    if (step_ptr) {  // hipGraph path: device-resident step counter
        const float st = (float)*step_ptr;
        bc1 = 1.0f - __powf(beta1, st);
        bc2 = 1.0f - __powf(beta2, st);
    }
    const int64_t stride = (int64_t)gridDim.x * 256;
    for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n; i += stride) {
        const float gi = g[i];
        const float mi = beta1 * m[i] + (1.0f - beta1) * gi;
        const float vi = beta2 * v[i] + (1.0f - beta2) * gi * gi;
        m[i] = mi;
        v[i] = vi;
        const float mhat = mi / bc1;
        const float vhat = vi / bc2;
        p[i] -= lr * mhat / (sqrtf(vhat) + eps);
    }
}

void adam_step(float* p, const float* g, float* m, float* v, int64_t n,
               float lr, float beta1, float beta2, float eps, int step,
               hipStream_t stream, const int* step_ptr) {
    const float bc1 = 1.0f - powf(beta1, float(step));
    const float bc2 = 1.0f - powf(beta2, float(step));
    int blocks = int((n + 255) / 256);
    if (blocks > 2048) blocks = 2048;
    hipLaunchKernelGGL(adam_step_kernel, dim3(blocks), dim3(256), 0, stream, p,
                       g, m, v, n, lr, beta1, beta2, eps, bc1, bc2, step_ptr);
}

// ---------------------------------------------------------------------------
// multi-tensor variants: gradients live in per-parameter tensors (autograd
// with grad=None assignment — no AccumulateGrad add kernel per parameter,
// which cost ~123 us/step as 26 tiny aten launches). `table` rows are
// [grad_ptr, flat_offset, numel]; params/moments stay in the flat buffers.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void adam_mt_kernel(
    const int64_t* __restrict__ table, int n_params, float* __restrict__ p,
    float* __restrict__ m, float* __restrict__ v, float lr, float beta1,
    float beta2, float eps, float bc1, float bc2,
    const int* __restrict__ step_ptr) {  // hipGraph path: step counter lives
                                         // on device, bias corrections are
                                         // computed in-kernel
    const int pi = blockIdx.y;
    if (pi >= n_params) return;
    if (step_ptr) {
        const float st = (float)*step_ptr;
        bc1 = 1.0f - __powf(beta1, st);
        bc2 = 1.0f - __powf(beta2, st);
    }
    const float* g = reinterpret_cast<const float*>(table[pi * 3 + 0]);
    const int64_t off = table[pi * 3 + 1];
    const int64_t n = table[pi * 3 + 2];
    const int64_t stride = (int64_t)gridDim.x * 256;
    for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n;
         i += stride) {
        const float gi = g[i];
        const int64_t k = off + i;
        const float mi = beta1 * m[k] + (1.0f - beta1) * gi;
        const float vi = beta2 * v[k] + (1.0f - beta2) * gi * gi;
        m[k] = mi;
        v[k] = vi;
        p[k] -= lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
    }
}

// gather per-tensor grads into the flat buffer (DP path: one launch, then
// one all-reduce on the flat tensor, then the flat adam_step)
__global__ __launch_bounds__(256) void grad_gather_kernel(
    const int64_t* __restrict__ table, int n_params,
    float* __restrict__ flat_g) {
    const int pi = blockIdx.y;
    if (pi >= n_params) return;
    const float* g = reinterpret_cast<const float*>(table[pi * 3 + 0]);
    const int64_t off = table[pi * 3 + 1];
    const int64_t n = table[pi * 3 + 2];
    const int64_t stride = (int64_t)gridDim.x * 256;
    for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n;
         i += stride) flat_g[off + i] = g[i];
}

void adam_mt(const int64_t* table, int n_params, float* p, float* m, float* v,
             float lr, float beta1, float beta2, float eps, int step,
             hipStream_t stream, const int* step_ptr) {
    const float bc1 = 1.0f - powf(beta1, float(step));
    const float bc2 = 1.0f - powf(beta2, float(step));
    hipLaunchKernelGGL(adam_mt_kernel, dim3(32, n_params), dim3(256), 0,
                       stream, table, n_params, p, m, v, lr, beta1, beta2, eps,
                       bc1, bc2, step_ptr);
}

void grad_gather(const int64_t* table, int n_params, float* flat_g,
                 hipStream_t stream) {
    hipLaunchKernelGGL(grad_gather_kernel, dim3(32, n_params), dim3(256), 0,
                       stream, table, n_params, flat_g);
}

}  // namespace rk
