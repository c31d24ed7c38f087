// Fused TRAINING front: embedding gather + dropout + fc1 + relu + dropout +
// fc2 + relu + dropout, forward AND backward, for gfx950.
//
// Replaces the reference's rnn_model.py:47-56 training graph (embedding ->
// permute -> fc1 -> fc2 -> reshape) which, executed op-by-op on torch/aten,
// cost ~3 ms/step at b=128 in dropout/reduce/scatter/GEMM glue
// (profiles/train_r01_kernel_stats.txt). Design:
//
//   * one workgroup (8 waves) per window; all weights staged in LDS once;
//     the 90 columns are processed sequentially, each as two small MFMA
//     GEMMs (t1 = relu(W1·m + b1), t2 = relu(W2·t1 + b2)) over LDS tiles;
//   * dropout masks are COUNTER-BASED HASHES (common.h drop_keep) of
//     (seed, element index) — nothing is materialised; the backward kernel
//     regenerates the embedding mask from the same seed and recovers the
//     relu/dropout deriatives of t1/t2 from the recomputed activations'
//     signs (post > 0 <=> kept AND pre > 0);
//   * the backward RECOMPUTES m and t1 per column (MFMA is ~100x cheaper
//     than round-tripping the 173 MB of activations through HBM), carries
//     dW1/dW2/db1/db2 in REGISTER fragment accumulators across all 90
//     columns, accumulates the embedding gradient in a (12,50) LDS table,
//     and commits everything with one atomic pass per workgroup at the end;
//   * dx is never needed: the input is integer base ids.
//
// Weight-layout notes: B-operand fragments are read either from transposed
// LDS tiles (ds_read_b128, lds_load_b_frag_t) or DIRECTLY from row-major
// tiles with 8 scalar reads (lds_load_b_frag_km) where a second transposed
// copy would not fit the 160 KB LDS budget.

#include <cstdint>

#include "common.h"

namespace rk {
namespace front {

constexpr int R = 200;    // reads per window
constexpr int W = 90;     // columns per window
constexpr int E = 50;     // embedding dim
constexpr int F1 = 100;   // fc1 out
constexpr int F2 = 10;    // fc2 out
constexpr int OUT = 500;  // E * F2

constexpr int KP = 224;        // R padded to 7 k-steps of 32
constexpr int KP_LD = KP + 8;  // +8 bf16 padding against bank conflicts
constexpr int MP = 112;        // F1 padded to 7 m-tiles
constexpr int EP = 64;         // E padded to 4 n-tiles

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(512, 2) void front_fwd_kernel(
    const uint8_t* __restrict__ ids,  // (B, R, W)
    const bf16* __restrict__ w1,      // (F1, R)
    const float* __restrict__ b1,     // (F1)
    const bf16* __restrict__ w2,      // (F2, F1)
    const float* __restrict__ b2,     // (F2)
    const bf16* __restrict__ emb,     // (12, E)
    bf16* __restrict__ out,           // (W, B, OUT)
    int B, uint32_t seed, float keep) {
    __shared__ struct {
        bf16 w1t[MP][KP_LD];     // zero-padded W1 [f][r]
        bf16 m_t[EP][KP_LD];     // masked embedding tile [e][r]
        bf16 t1_t[EP][136];      // t1 post-activation [e][f]
        bf16 w2_lds[16][136];    // zero-padded W2 [j][f]
        bf16 emb_s[12][E];
        bf16 t2st[OUT + 12];
        float b1s[MP];
        float b2s[16];
        uint8_t col_ids[R];
    } lds;

    const int b = blockIdx.x;
    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int lrow = lane >> 4;
    const int lcol = lane & 15;
    const uint64_t thresh = (uint64_t)((double)keep * 4294967296.0);
    const float inv_keep = 1.0f / keep;

    // ---- one-time staging -------------------------------------------------
    for (int e = tid; e < MP * KP_LD; e += 512) (&lds.w1t[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < EP * KP_LD; e += 512) (&lds.m_t[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < EP * 136; e += 512) (&lds.t1_t[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < 16 * 136; e += 512) (&lds.w2_lds[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < MP; e += 512) lds.b1s[e] = 0.f;
    for (int e = tid; e < 16; e += 512) lds.b2s[e] = 0.f;
    __syncthreads();
    for (int e = tid; e < F1 * R; e += 512) lds.w1t[e / R][e % R] = w1[e];
    for (int e = tid; e < 12 * E; e += 512) lds.emb_s[e / E][e % E] = emb[e];
    for (int e = tid; e < F2 * F1; e += 512) lds.w2_lds[e / F1][e % F1] = w2[e];
    for (int e = tid; e < F1; e += 512) lds.b1s[e] = b1[e];
    for (int e = tid; e < F2; e += 512) lds.b2s[e] = b2[e];
    __syncthreads();

    for (int w = 0; w < W; ++w) {
        // ---- stage this column's read ids --------------------------------
        if (tid < R) lds.col_ids[tid] = ids[((size_t)b * R + tid) * W + w];
        __syncthreads();
        // ---- masked embedding tile m[r][e] stored [e][r] ------------------
        for (int i = tid; i < R * E; i += 512) {
            const int r = i / E, e = i % E;
            const uint32_t idx = ((uint32_t)(b * R + r) * W + w) * E + e;
            float v = 0.f;
            if (drop_keep(seed, idx, thresh))
                v = bf2f(lds.emb_s[lds.col_ids[r]][e]) * inv_keep;
            lds.m_t[e][r] = f2bf(v);
        }
        __syncthreads();

        // ---- G1: t1 = drop(relu(W1 · m + b1)) — 28 tiles over 8 waves ----
#pragma unroll
        for (int s = 0; s < 4; ++s) {
            const int tile = wid + s * 8;
            if (tile < 28) {
                const int mt = tile >> 2, nt = tile & 3;
                f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
                for (int kb = 0; kb < 7; ++kb) {
                    bf16x8 a = lds_load_a_frag(&lds.w1t[0][0], mt * 16, kb * 32, KP_LD);
                    bf16x8 bb = lds_load_b_frag_t(&lds.m_t[0][0], nt * 16, kb * 32, KP_LD);
                    acc = mfma16x16x32(a, bb, acc);
                }
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    const int f = mt * 16 + lrow * 4 + i;
                    const int e = nt * 16 + lcol;
                    float v = fmaxf(acc[i] + lds.b1s[f], 0.f);
                    const uint32_t idx = ((uint32_t)(b * W + w) * F1 + f) * E + e;
                    const bool live = (f < F1) && (e < E) &&
                                      drop_keep(seed ^ 0x51u, idx, thresh);
                    lds.t1_t[e][f] = f2bf(live ? v * inv_keep : 0.f);
                }
            }
        }
        __syncthreads();

        // ---- G3: t2 = drop(relu(W2 · t1 + b2)) — 4 tiles, waves 0..3 -----
        if (wid < 4) {
            const int nt = wid;
            f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int kb = 0; kb < 4; ++kb) {
                bf16x8 a = lds_load_a_frag(&lds.w2_lds[0][0], 0, kb * 32, 136);
                bf16x8 bb = lds_load_b_frag_t(&lds.t1_t[0][0], nt * 16, kb * 32, 136);
                acc = mfma16x16x32(a, bb, acc);
            }
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int j = lrow * 4 + i;
                const int e = nt * 16 + lcol;
                if (j < F2 && e < E) {
                    float v = fmaxf(acc[i] + lds.b2s[j], 0.f);
                    const uint32_t idx = ((uint32_t)(b * W + w) * F2 + j) * E + e;
                    const bool live = drop_keep(seed ^ 0x52u, idx, thresh);
                    lds.t2st[e * F2 + j] = f2bf(live ? v * inv_keep : 0.f);
                }
            }
        }
        __syncthreads();

        // ---- coalesced store: out[(w, b), :] ------------------------------
        {
            bf16* dst = out + ((size_t)w * B + b) * OUT;
            if (tid < 62)
                *reinterpret_cast<bf16x8*>(dst + tid * 8) =
                    *reinterpret_cast<const bf16x8*>(&lds.t2st[tid * 8]);
            else if (tid == 62)
#pragma unroll
                for (int q = 0; q < 4; ++q) dst[496 + q] = lds.t2st[496 + q];
        }
        __syncthreads();
    }
}

// ---------------------------------------------------------------------------
// backward (recompute)
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(512, 2) void front_bwd_kernel(
    const uint8_t* __restrict__ ids,   // (B, R, W)
    const bf16* __restrict__ dseq,     // (W, B, OUT) grad wrt forward output
    const bf16* __restrict__ w1, const float* __restrict__ b1,
    const bf16* __restrict__ w2, const float* __restrict__ b2,
    const bf16* __restrict__ emb,
    float* __restrict__ dw1,  // (F1, R)  pre-zeroed, atomic-accumulated
    float* __restrict__ db1,  // (F1)
    float* __restrict__ dw2,  // (F2, F1)
    float* __restrict__ db2,  // (F2)
    float* __restrict__ de,   // (12, E)
    int B, uint32_t seed, float keep) {
    __shared__ struct {
        bf16 w1t[128][KP_LD];    // zero-padded W1 [f][r] (128 rows: K reads)
        bf16 m_t[EP][KP_LD];     // [e][r]
        bf16 t1_t[EP][136];      // [e][f] recomputed t1 post
        bf16 w2_lds[16][136];    // [j][f]
        bf16 w2t_t[MP][40];      // [f][j] = W2^T zero-padded
        bf16 dt2_je[32][72];     // [j][e] dt2 pre-activation grads
        bf16 dt1_fe[128][72];    // [f][e] dt1 pre-activation grads
        bf16 dseq_st[OUT + 12];
        bf16 emb_s[12][E];
        float b1s[MP];
        float b2s[16];
        float de_lds[12][E];
        float db1_lds[MP];
        float db2_lds[16];
        uint8_t col_ids[R];
    } lds;

    const int b = blockIdx.x;
    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int lrow = lane >> 4;
    const int lcol = lane & 15;
    const uint64_t thresh = (uint64_t)((double)keep * 4294967296.0);
    const float inv_keep = 1.0f / keep;

    // ---- one-time staging + zero ------------------------------------------
    for (int e = tid; e < 128 * KP_LD; e += 512) (&lds.w1t[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < EP * KP_LD; e += 512) (&lds.m_t[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < EP * 136; e += 512) (&lds.t1_t[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < 16 * 136; e += 512) (&lds.w2_lds[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < MP * 40; e += 512) (&lds.w2t_t[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < 32 * 72; e += 512) (&lds.dt2_je[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < 128 * 72; e += 512) (&lds.dt1_fe[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < 12 * E; e += 512) (&lds.de_lds[0][0])[e] = 0.f;
    for (int e = tid; e < MP; e += 512) lds.db1_lds[e] = 0.f;
    for (int e = tid; e < 16; e += 512) lds.db2_lds[e] = 0.f;
    for (int e = tid; e < MP; e += 512) lds.b1s[e] = 0.f;
    for (int e = tid; e < 16; e += 512) lds.b2s[e] = 0.f;
    __syncthreads();
    for (int e = tid; e < F1 * R; e += 512) lds.w1t[e / R][e % R] = w1[e];
    for (int e = tid; e < 12 * E; e += 512) lds.emb_s[e / E][e % E] = emb[e];
    for (int e = tid; e < F2 * F1; e += 512) {
        lds.w2_lds[e / F1][e % F1] = w2[e];
        lds.w2t_t[e % F1][e / F1] = w2[e];
    }
    for (int e = tid; e < F1; e += 512) lds.b1s[e] = b1[e];
    for (int e = tid; e < F2; e += 512) lds.b2s[e] = b2[e];
    __syncthreads();

    // register accumulators carried across all 90 columns
    f32x4 dw1acc[12];
#pragma unroll
    for (int s = 0; s < 12; ++s) dw1acc[s] = f32x4{0.f, 0.f, 0.f, 0.f};
    f32x4 dw2acc = {0.f, 0.f, 0.f, 0.f};

    for (int w = 0; w < W; ++w) {
        // ---- stage ids column + dseq column ------------------------------
        if (tid < R) lds.col_ids[tid] = ids[((size_t)b * R + tid) * W + w];
        if (tid < OUT)
            lds.dseq_st[tid] = dseq[((size_t)w * B + b) * OUT + tid];
        __syncthreads();
        // ---- recompute masked embedding tile ------------------------------
        for (int i = tid; i < R * E; i += 512) {
            const int r = i / E, e = i % E;
            const uint32_t idx = ((uint32_t)(b * R + r) * W + w) * E + e;
            float v = 0.f;
            if (drop_keep(seed, idx, thresh))
                v = bf2f(lds.emb_s[lds.col_ids[r]][e]) * inv_keep;
            lds.m_t[e][r] = f2bf(v);
        }
        __syncthreads();

        // ---- recompute t1 (same code path as forward => same bf16 bits) ---
#pragma clang loop unroll(disable)
        for (int s = 0; s < 4; ++s) {
            const int tile = wid + s * 8;
            if (tile < 28) {
                const int mt = tile >> 2, nt = tile & 3;
                f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
                for (int kb = 0; kb < 7; ++kb) {
                    bf16x8 a = lds_load_a_frag(&lds.w1t[0][0], mt * 16, kb * 32, KP_LD);
                    bf16x8 bb = lds_load_b_frag_t(&lds.m_t[0][0], nt * 16, kb * 32, KP_LD);
                    acc = mfma16x16x32(a, bb, acc);
                }
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    const int f = mt * 16 + lrow * 4 + i;
                    const int e = nt * 16 + lcol;
                    float v = fmaxf(acc[i] + lds.b1s[f], 0.f);
                    const uint32_t idx = ((uint32_t)(b * W + w) * F1 + f) * E + e;
                    const bool live = (f < F1) && (e < E) &&
                                      drop_keep(seed ^ 0x51u, idx, thresh);
                    lds.t1_t[e][f] = f2bf(live ? v * inv_keep : 0.f);
                }
            }
        }
        __syncthreads();

        // ---- recompute t2pre; dt2 = dseq ⊙ drop2' ⊙ relu2' ---------------
        if (wid < 4) {
            const int nt = wid;
            f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int kb = 0; kb < 4; ++kb) {
                bf16x8 a = lds_load_a_frag(&lds.w2_lds[0][0], 0, kb * 32, 136);
                bf16x8 bb = lds_load_b_frag_t(&lds.t1_t[0][0], nt * 16, kb * 32, 136);
                acc = mfma16x16x32(a, bb, acc);
            }
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int j = lrow * 4 + i;
                const int e = nt * 16 + lcol;
                if (j < F2 && e < E) {
                    const float t2pre = acc[i] + lds.b2s[j];
                    const uint32_t idx = ((uint32_t)(b * W + w) * F2 + j) * E + e;
                    const bool live = (t2pre > 0.f) &&
                                      drop_keep(seed ^ 0x52u, idx, thresh);
                    const float g = live
                        ? bf2f(lds.dseq_st[e * F2 + j]) * inv_keep : 0.f;
                    lds.dt2_je[j][e] = f2bf(g);
                    float red = g;
#pragma unroll
                    for (int moff = 1; moff < 16; moff <<= 1)
                        red += __shfl_xor(red, moff, 16);
                    if (lcol == 0) atomicAdd(&lds.db2_lds[j], red);
                }
            }
        }
        __syncthreads();

        // ---- dt1 = W2^T · dt2, through relu1'/drop1' ----------------------
#pragma clang loop unroll(disable)
        for (int s = 0; s < 4; ++s) {
            const int tile = wid + s * 8;
            if (tile < 28) {
                const int mt = tile >> 2, nt = tile & 3;
                bf16x8 a = lds_load_a_frag(&lds.w2t_t[0][0], mt * 16, 0, 40);
                bf16x8 bb = lds_load_b_frag_km(&lds.dt2_je[0][0], 0, nt * 16, 72);
                f32x4 acc = mfma16x16x32(a, bb, f32x4{0.f, 0.f, 0.f, 0.f});
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    const int f = mt * 16 + lrow * 4 + i;
                    const int e = nt * 16 + lcol;
                    // post > 0 <=> kept AND pre > 0 (chain rule collapses)
                    const float t1post = bf2f(lds.t1_t[e][f]);
                    const float g = (t1post > 0.f) ? acc[i] * inv_keep : 0.f;
                    lds.dt1_fe[f][e] = f2bf(g);
                    float red = g;
#pragma unroll
                    for (int moff = 1; moff < 16; moff <<= 1)
                        red += __shfl_xor(red, moff, 16);
                    if (lcol == 0) atomicAdd(&lds.db1_lds[f], red);
                }
            }
        }
        __syncthreads();

        // ---- read-only phase: dW2, dW1 accumulate; dm -> de ---------------
        if (wid < 7) {  // dW2 += dt2 · t1^T   (A (j,e), B (e,f))
            const int nt = wid;
#pragma unroll
            for (int kb = 0; kb < 2; ++kb) {
                bf16x8 a = lds_load_a_frag(&lds.dt2_je[0][0], 0, kb * 32, 72);
                bf16x8 bb = lds_load_b_frag_km(&lds.t1_t[0][0], kb * 32, nt * 16, 136);
                dw2acc = mfma16x16x32(a, bb, dw2acc);
            }
        }
#pragma unroll
        for (int s = 0; s < 12; ++s) {  // dW1 += dt1 · m^T  (A (f,e), B (e,r))
            const int tile = wid + s * 8;
            if (tile < 91) {
                const int mt = tile / 13, nt = tile % 13;
#pragma unroll
                for (int kb = 0; kb < 2; ++kb) {
                    bf16x8 a = lds_load_a_frag(&lds.dt1_fe[0][0], mt * 16, kb * 32, 72);
                    bf16x8 bb = lds_load_b_frag_km(&lds.m_t[0][0], kb * 32, nt * 16, KP_LD);
                    dw1acc[s] = mfma16x16x32(a, bb, dw1acc[s]);
                }
            }
        }
#pragma clang loop unroll(disable)
        for (int s = 0; s < 7; ++s) {  // dm = W1^T · dt1  (A (r,f), B (f,e))
            const int tile = wid + s * 8;
            if (tile < 52) {
                const int mt = tile >> 2, nt = tile & 3;
                f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
                for (int kb = 0; kb < 4; ++kb) {
                    bf16x8 a = lds_load_a_frag_t(&lds.w1t[0][0], mt * 16, kb * 32, KP_LD);
                    bf16x8 bb = lds_load_b_frag_km(&lds.dt1_fe[0][0], kb * 32, nt * 16, 72);
                    acc = mfma16x16x32(a, bb, acc);
                }
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    const int r = mt * 16 + lrow * 4 + i;
                    const int e = nt * 16 + lcol;
                    if (r < R && e < E) {
                        const uint32_t idx = ((uint32_t)(b * R + r) * W + w) * E + e;
                        if (drop_keep(seed, idx, thresh))
                            atomicAdd(&lds.de_lds[lds.col_ids[r]][e],
                                      acc[i] * inv_keep);
                    }
                }
            }
        }
        __syncthreads();
    }

    // ---- commit the per-workgroup accumulators ----------------------------
#pragma unroll
    for (int s = 0; s < 12; ++s) {
        const int tile = wid + s * 8;
        if (tile < 91) {
            const int mt = tile / 13, nt = tile % 13;
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int f = mt * 16 + lrow * 4 + i;
                const int r = nt * 16 + lcol;
                if (f < F1 && r < R) atomicAdd(&dw1[f * R + r], dw1acc[s][i]);
            }
        }
    }
    if (wid < 7) {
        const int nt = wid;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int j = lrow * 4 + i;
            const int f = nt * 16 + lcol;
            if (j < F2 && f < F1) atomicAdd(&dw2[j * F1 + f], dw2acc[i]);
        }
    }
    __syncthreads();
    for (int i = tid; i < F1; i += 512)
        if (lds.db1_lds[i] != 0.f) atomicAdd(&db1[i], lds.db1_lds[i]);
    for (int i = tid; i < F2; i += 512)
        if (lds.db2_lds[i] != 0.f) atomicAdd(&db2[i], lds.db2_lds[i]);
    for (int i = tid; i < 12 * E; i += 512)
        if ((&lds.de_lds[0][0])[i] != 0.f)
            atomicAdd(&de[i], (&lds.de_lds[0][0])[i]);
}

}  // namespace front

void front_fwd(const uint8_t* ids, const void* w1, const float* b1,
               const void* w2, const float* b2, const void* emb, void* out,
               int B, uint32_t seed, float keep, hipStream_t stream) {
    hipLaunchKernelGGL(front::front_fwd_kernel, dim3(B), dim3(512), 0, stream,
                       ids, static_cast<const bf16*>(w1), b1,
                       static_cast<const bf16*>(w2), b2,
                       static_cast<const bf16*>(emb), static_cast<bf16*>(out),
                       B, seed, keep);
}

void front_bwd(const uint8_t* ids, const void* dseq, const void* w1,
               const float* b1, const void* w2, const float* b2,
               const void* emb, float* dw1, float* db1, float* dw2, float* db2,
               float* de, int B, uint32_t seed, float keep,
               hipStream_t stream) {
    hipLaunchKernelGGL(front::front_bwd_kernel, dim3(B), dim3(512), 0, stream,
                       ids, static_cast<const bf16*>(dseq),
                       static_cast<const bf16*>(w1), b1,
                       static_cast<const bf16*>(w2), b2,
                       static_cast<const bf16*>(emb), dw1, db1, dw2, db2, de,
                       B, seed, keep);
}

}  // namespace rk
