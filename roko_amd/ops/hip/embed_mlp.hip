// Fused embedding + per-column MLP reduction forward (inference/eval).
//
// Replaces the reference's embedding -> fc1 -> fc2 chain
// (rnn_model.py:47-56). Key algebraic re-design for
// CDNA4 instead of a translation: since the embedding table has only 12
// rows, the read-axis reduction factors through the base classes:
//     t1 = relu(W1 · E[ids] + b1)  ==  relu(W1 · (Hot · E) + b1)
//                                  ==  relu((W1 · Hot) · E + b1)
// with Hot the 200x12 one-hot matrix of the column's read bases. This cuts
// the per-column MAC count ~8x (200->12 contraction) and turns the gather
// into a 200-entry scatter of ones. Exact in eval mode (no dropout).
//
// One workgroup (EIGHT waves, two per SIMD) owns one window and processes
// TWO columns per iteration: waves 0-3 run column w, waves 4-7 column w+1,
// with private hot/a/t1/t2 tiles per column group and the weights staged
// once in shared LDS. The kernel is latency-bound, not MFMA-bound (~6k
// cycles per column against ~400 cycles of MFMA issue), so the win is the
// CDNA4 arbitration rule (docs/KERNELS.md "Measured lessons" #1): a
// co-resident partner wave's MFMAs fill the stall slots of the other
// column's dependent LDS/VALU chains. Each column is three small MFMA GEMMs:
//   G1: A   (100x12) = W1 (100x200) · Hot (200x12)        49 MFMA
//   G2: t1  (100x50) = relu(A · E (12x50) + b1)           28 MFMA
//   G3: t2  (10x50)  = relu(W2 (10x100) · t1 + b2)        16 MFMA
// out[(w,b), e*10+j] = t2[j][e]  ->  (T=90, B, 500) bf16 for the GRU GEMM.

#include <cstdint>

#include "common.h"

namespace rk {

constexpr int R = 200;    // reads per window
constexpr int W = 90;     // columns per window
constexpr int E = 50;     // embedding dim
constexpr int F1 = 100;   // fc1 out
constexpr int F2 = 10;    // fc2 out
constexpr int OUT = 500;  // E * F2

constexpr int KP = 224;           // R padded to 7 k-steps of 32
constexpr int KP_LD = KP + 8;     // +8 bf16 padding against bank conflicts
constexpr int MP = 112;           // F1 padded to 7 m-tiles

__global__ __launch_bounds__(512, 2) void embed_mlp_fwd_kernel(
    const uint8_t* __restrict__ ids,  // (B, R, W)
    const bf16* __restrict__ w1,      // (F1, R)
    const float* __restrict__ b1,     // (F1)
    const bf16* __restrict__ w2,      // (F2, F1)
    const float* __restrict__ b2,     // (F2)
    const bf16* __restrict__ emb,     // (12, E)
    bf16* __restrict__ out,           // (W, B, OUT)
    int B) {
    __shared__ struct {
        // shared across both column groups (read-only after staging)
        bf16 w1t[MP][KP_LD];        // A-operand of G1 (zero-padded W1)
        uint8_t win[R * W];         // the window
        bf16 e_t[64][32];           // B-operand of G2, [col e][k c]
        bf16 w2_lds[16][136];       // A-operand of G3 (zero-padded W2)
        float b1s[F1];
        float b2s[F2];
        // per column group g = 0/1
        bf16 hot_t[2][16][KP_LD];   // B-operand of G1, stored [col][k]
        bf16 a_lds[2][MP][40];      // A (G1 out / G2 A-op), k-pad to 32(+8)
        bf16 t1_t[2][64][136];      // B-operand of G3, [col e][k f] (+pad)
        bf16 t2[2][512];            // staging for the coalesced output store
    } lds;

    const int b = blockIdx.x;
    const int tid = threadIdx.x;
    const int g = tid >> 8;          // column group (waves 0-3 / 4-7)
    const int tidg = tid & 255;      // thread id within the group
    const int wid4 = (tid >> 6) & 3; // wave id within the group
    const int lane = tid & 63;
    const int lrow = lane >> 4;      // fragment row group (0..3)
    const int lcol = lane & 15;      // fragment column

    // ---- one-time staging (all 512 threads) -------------------------------
    for (int e = tid; e < MP * KP_LD; e += 512) (&lds.w1t[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < 64 * 32; e += 512) (&lds.e_t[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < 2 * 64 * 136; e += 512)
        (&lds.t1_t[0][0][0])[e] = f2bf(0.f);
    for (int e = tid; e < 16 * 136; e += 512) (&lds.w2_lds[0][0])[e] = f2bf(0.f);
    __syncthreads();
    for (int e = tid; e < F1 * R; e += 512) lds.w1t[e / R][e % R] = w1[e];
    for (int e = tid; e < 12 * E; e += 512) lds.e_t[e % E][e / E] = emb[e];
    for (int e = tid; e < F2 * F1; e += 512) lds.w2_lds[e / F1][e % F1] = w2[e];
    for (int e = tid; e < F1; e += 512) lds.b1s[e] = b1[e];
    for (int e = tid; e < F2; e += 512) lds.b2s[e] = b2[e];
    {
        const uint8_t* src = ids + (size_t)b * R * W;
        for (int e = tid * 8; e < R * W; e += 512 * 8)
#pragma unroll
            for (int q = 0; q < 8 && e + q < R * W; ++q) lds.win[e + q] = src[e + q];
    }
    __syncthreads();

    // ---- per-column-pair loop (group g owns column wp + g) ----------------
    for (int wp = 0; wp < W; wp += 2) {
        const int w = wp + g;
        // build Hot^T: zero, then scatter 200 ones
        for (int e = tidg; e < 16 * KP_LD; e += 256)
            (&lds.hot_t[g][0][0])[e] = f2bf(0.f);
        __syncthreads();
        for (int r = tidg; r < R; r += 256)
            lds.hot_t[g][lds.win[r * W + w]][r] = f2bf(1.0f);
        __syncthreads();

        // G1: A = W1 · Hot  — the group's waves own m-tiles {wid4, wid4+4}
#pragma unroll
        for (int s = 0; s < 2; ++s) {
            const int mt = wid4 + s * 4;
            if (mt < 7) {
                f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
                for (int kb = 0; kb < 7; ++kb) {
                    bf16x8 a = lds_load_a_frag(&lds.w1t[0][0], mt * 16, kb * 32, KP_LD);
                    bf16x8 bb = lds_load_b_frag_t(&lds.hot_t[g][0][0], 0, kb * 32, KP_LD);
                    acc = mfma16x16x32(a, bb, acc);
                }
#pragma unroll
                for (int i = 0; i < 4; ++i)
                    lds.a_lds[g][mt * 16 + lrow * 4 + i][lcol] = f2bf(acc[i]);
            }
        }
        // zero the k-pad of a_lds once: cols 12..15 are written with
        // zero-valued products (Hot cols 12..15 are zero); cols 16..31 are
        // cleared here on the first iteration only
        if (wp == 0)
            for (int e = tidg; e < MP; e += 256)
#pragma unroll
                for (int k = 16; k < 32; ++k) lds.a_lds[g][e][k] = f2bf(0.f);
        __syncthreads();

        // G2: t1 = relu(A · E + b1) — 7 m-tiles x 4 n-tiles over 4 waves
#pragma unroll
        for (int s = 0; s < 7; ++s) {
            const int tile = wid4 + s * 4;
            if (tile < 28) {
                const int mt = tile >> 2, nt = tile & 3;
                bf16x8 a = lds_load_a_frag(&lds.a_lds[g][0][0], mt * 16, 0, 40);
                bf16x8 bb = lds_load_b_frag_t(&lds.e_t[0][0], nt * 16, 0, 32);
                f32x4 acc = mfma16x16x32(a, bb, f32x4{0.f, 0.f, 0.f, 0.f});
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    const int f = mt * 16 + lrow * 4 + i;
                    const int e = nt * 16 + lcol;
                    float v = acc[i] + (f < F1 ? lds.b1s[f] : 0.f);
                    lds.t1_t[g][e][f] = f2bf(fmaxf(v, 0.f));
                }
            }
        }
        __syncthreads();

        // G3: t2 = relu(W2 · t1 + b2) — 4 n-tiles, one per wave
        {
            const int nt = wid4;
            f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int kb = 0; kb < 4; ++kb) {
                bf16x8 a = lds_load_a_frag(&lds.w2_lds[0][0], 0, kb * 32, 136);
                bf16x8 bb = lds_load_b_frag_t(&lds.t1_t[g][0][0], nt * 16, kb * 32, 136);
                acc = mfma16x16x32(a, bb, acc);
            }
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int j = lrow * 4 + i;
                const int e = nt * 16 + lcol;
                if (j < F2 && e < E) {
                    float v = acc[i] + lds.b2s[j];
                    lds.t2[g][e * F2 + j] = f2bf(fmaxf(v, 0.f));
                }
            }
        }
        __syncthreads();

        // coalesced store: out[(w, b), :] = t2 flattened (e*10+j)
        {
            bf16* dst = out + ((size_t)w * B + b) * OUT;
            if (tidg < 62)
                *reinterpret_cast<bf16x8*>(dst + tidg * 8) =
                    *reinterpret_cast<const bf16x8*>(&lds.t2[g][tidg * 8]);
            else if (tidg == 62)
#pragma unroll
                for (int q = 0; q < 4; ++q) dst[496 + q] = lds.t2[g][496 + q];
        }
        __syncthreads();
    }
}

void embed_mlp_fwd(const uint8_t* ids, const void* w1, const float* b1,
                   const void* w2, const float* b2, const void* emb, void* out,
                   int B, hipStream_t stream) {
    hipLaunchKernelGGL(embed_mlp_fwd_kernel, dim3(B), dim3(512), 0, stream,
                       ids, static_cast<const bf16*>(w1), b1,
                       static_cast<const bf16*>(w2), b2,
                       static_cast<const bf16*>(emb), static_cast<bf16*>(out), B);
}

}  // namespace rk
