// MFMA fragment-layout probe: computes one 16x16x32 bf16 MFMA through the
// exact fragment loaders the real kernels use (common.h), so a host-side
// comparison against a plain matmul validates the lane->element mappings
// before anything else trusts them (cdna_hip_programming.md §3: always
// verify with ASYMMETRIC operands).

#include "common.h"

namespace rk {

__global__ void mfma_probe_kernel(const float* __restrict__ a_in,  // (16,32)
                                  const float* __restrict__ b_in,  // (32,16)
                                  float* __restrict__ d_out) {     // (16,16)
    __shared__ struct {
        bf16 a[16][40];   // row-major, padded
        bf16 bt[16][40];  // B stored transposed [col][k]
    } lds;
    const int lane = threadIdx.x & 63;
    if (threadIdx.x < 64) {
        for (int e = lane; e < 16 * 32; e += 64)
            lds.a[e / 32][e % 32] = f2bf(a_in[e]);
        // b_in is (32,16) row-major: element (k, col) at k*16+col
        for (int k = lane; k < 32; k += 64)
            for (int c = 0; c < 16; ++c) lds.bt[c][k] = f2bf(b_in[k * 16 + c]);
    }
    __syncthreads();
    if (threadIdx.x < 64) {
        bf16x8 a = lds_load_a_frag(&lds.a[0][0], 0, 0, 40);
        bf16x8 b = lds_load_b_frag_t(&lds.bt[0][0], 0, 0, 40);
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        acc = mfma16x16x32(a, b, acc);
        const int col = lane & 15, rowg = lane >> 4;
#pragma unroll
        for (int i = 0; i < 4; ++i) d_out[(rowg * 4 + i) * 16 + col] = acc[i];
    }
}

void mfma_probe(const float* a, const float* b, float* d, hipStream_t stream) {
    hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream, a, b, d);
}

}  // namespace rk
