// Shared helpers for the CDNA4 (gfx950) kernels. HIP-only — no CUDA compat.
//
// MFMA note: all matrix math uses v_mfma_f32_16x16x32_bf16 (gfx950's 2xK
// bf16 shape, cdna_hip_programming.md §3). Fragment mappings used here:
//   A (16x32):  lane l holds A[row = l & 15][k = 8*(l >> 4) + j], j = 0..7
//   B (32x16):  lane l holds B[k = 8*(l >> 4) + j][col = l & 15]
//   C/D (16x16): lane l, reg i -> D[row = (l >> 4)*4 + i][col = l & 15]
// The mappings are verified on-device by the mfma_probe kernel (test_gpu.py)
// before anything else trusts them.

#pragma once

#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#define RK_DEV __device__ __forceinline__

namespace rk {

using bf16 = __hip_bfloat16;
using f32x4 = __attribute__((__vector_size__(4 * sizeof(float)))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using bf16x4 = __attribute__((ext_vector_type(4))) __bf16;
using short4_t = __attribute__((ext_vector_type(4))) short;

RK_DEV float bf2f(bf16 v) { return __bfloat162float(v); }
RK_DEV bf16 f2bf(float v) { return __float2bfloat16(v); }

// MFMA wrapper: D = A(16x32) * B(32x16) + C, bf16 in / f32 accumulate.
RK_DEV f32x4 mfma16x16x32(bf16x8 a, bf16x8 b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// Load an A-fragment from an LDS tile of bf16 with row stride `ld` elements:
// rows [row0, row0+16), k-columns [k0, k0+32).
RK_DEV bf16x8 lds_load_a_frag(const bf16* tile, int row0, int k0, int ld) {
    const int lane = threadIdx.x & 63;
    const int row = row0 + (lane & 15);
    const int k = k0 + 8 * (lane >> 4);
    return *reinterpret_cast<const bf16x8*>(tile + row * ld + k);
}

// Load a B-fragment B[k][col] from an LDS tile stored K-major (i.e. tile
// layout [col][k], row stride ld elements) — cols [col0, col0+16),
// k in [k0, k0+32). Reading B transposed-stored keeps lanes on 16B chunks.
RK_DEV bf16x8 lds_load_b_frag_t(const bf16* tile_t, int col0, int k0, int ld) {
    const int lane = threadIdx.x & 63;
    const int col = col0 + (lane & 15);
    const int k = k0 + 8 * (lane >> 4);
    return *reinterpret_cast<const bf16x8*>(tile_t + col * ld + k);
}

// B-fragment B[k][col] read DIRECTLY from a row-major [k][col] tile
// (8 scalar u16 reads per lane; use when a second transposed LDS copy of the
// tile would blow the LDS budget — the 2-cyc scalar issues hide under MFMAs).
RK_DEV bf16x8 lds_load_b_frag_km(const bf16* tile, int k0, int col0, int ld) {
    const int lane = threadIdx.x & 63;
    const int col = col0 + (lane & 15);
    const int kb = k0 + 8 * (lane >> 4);
    bf16x8 r;
#pragma unroll
    for (int j = 0; j < 8; ++j) r[j] = tile[(size_t)(kb + j) * ld + col];
    return r;
}

// A-fragment A[row][k] read from a TRANSPOSED [k][row] tile (8 scalar reads).
RK_DEV bf16x8 lds_load_a_frag_t(const bf16* tile_t, int row0, int k0, int ld) {
    const int lane = threadIdx.x & 63;
    const int row = row0 + (lane & 15);
    const int kb = k0 + 8 * (lane >> 4);
    bf16x8 r;
#pragma unroll
    for (int j = 0; j < 8; ++j) r[j] = tile_t[(size_t)(kb + j) * ld + row];
    return r;
}

// A-fragment loaded straight from a row-major GLOBAL tile (L2-resident
// weights): 16 B per lane, same mapping as lds_load_a_frag.
RK_DEV bf16x8 global_load_a_frag(const bf16* tile, int row0, int k0, int ld) {
    const int lane = threadIdx.x & 63;
    const int row = row0 + (lane & 15);
    const int k = k0 + 8 * (lane >> 4);
    return *reinterpret_cast<const bf16x8*>(tile + (size_t)row * ld + k);
}

// Counter-based dropout hash (lowbias32): deterministic per (seed, index),
// regenerated identically in the recompute backward — no mask tensors.
RK_DEV uint32_t hash32(uint32_t x) {
    x ^= x >> 16;
    x *= 0x7feb352dU;
    x ^= x >> 15;
    x *= 0x846ca68bU;
    x ^= x >> 16;
    return x;
}

// keep iff hash < thresh; thresh = keep_prob * 2^32 (uint64 so keep=1 -> all)
RK_DEV bool drop_keep(uint32_t seed, uint32_t idx, uint64_t thresh) {
    return (uint64_t)hash32(idx ^ (seed * 0x9E3779B9U)) < thresh;
}

// Relaxed atomic adds: plain atomicAdd defaults to seq_cst and hipcc fences
// EVERY ds_add_f32 behind an s_waitcnt lgkmcnt(0) — a full LDS-pipe drain
// per atomic (measured as the dominant cost of the de reduction). We only
// need atomicity, never ordering: results are read after kernel end (global)
// or after a __syncthreads() (LDS).
RK_DEV void lds_atomic_add(float* p, float v) {
    __hip_atomic_fetch_add(p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_WORKGROUP);
}
RK_DEV void agent_atomic_add(float* p, float v) {
    __hip_atomic_fetch_add(p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

// async global->LDS DMA, 16 B per lane (gfx950 global_load_lds_dwordx4).
// LDS destination is WAVE-UNIFORM base + lane*16 — the caller's LDS layout
// must be lane-linear; the global source address is per-lane.
RK_DEV void glds_b128(const void* g, void* l) {
#pragma clang diagnostic push
#pragma clang diagnostic ignored "-Wold-style-cast"
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)(g),
        (__attribute__((address_space(3))) uint32_t*)(l), 16, 0, 0);
#pragma clang diagnostic pop
}

// v_rcp_f32 (~1 ulp) instead of the IEEE division sequence: hipcc lowers
// `a / b` to div_scale/rcp/6xfma/div_fmas/div_fixup (~12 dependent VALU) —
// 24 of those per lane per GRU step measured as 55% of the whole recurrence
// kernel (scripts/gru_timing.py dbg=4 bisection). The 1-ulp rcp error is
// orders of magnitude below the bf16 state rounding.
RK_DEV float fast_rcp(float x) { return __builtin_amdgcn_rcpf(x); }

RK_DEV float sigmoidf_dev(float x) { return fast_rcp(1.0f + __expf(-x)); }

// overflow-stable tanh: tanh(x) = sign(x) * (1 - e) / (1 + e), e = exp(-2|x|)
RK_DEV float tanhf_dev(float x) {
    float e = __expf(-2.0f * fabsf(x));
    float t = (1.0f - e) * fast_rcp(1.0f + e);
    return copysignf(t, x);
}

#define HIP_CHECK_LAST()                                                     \
    do {                                                                     \
        hipError_t e_ = hipGetLastError();                                   \
        if (e_ != hipSuccess) {                                              \
            throw std::runtime_error(std::string("HIP kernel launch: ") +    \
                                     hipGetErrorString(e_));                 \
        }                                                                    \
    } while (0)

}  // namespace rk
