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
// TWO kernels live in this file. The first (v1) is the per-column version:
// one workgroup (eight waves, two per SIMD) owns one window, waves 0-3 run
// column w and waves 4-7 column w+1 with private hot/a/t1/t2 tiles. It is
// KEPT as the reference implementation and A/B fallback (ROKO_FRONT=v1):
// phase bisection + PMC showed it 49% issue-stall / 13% active, and neither
// the 2-column split, accumulator splitting nor load hoisting moved it —
// the cost is the barrier-fenced tiny-GEMM phase structure itself, which
// the chunked v2 kernel below replaces (1.85x). Each column is three small
// MFMA GEMMs:
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
    int B,
    uint32_t dbg,                     // phase-skip bisection: 1 hot, 2 G1,
                                      // 4 G2, 8 G3, 16 store
    unsigned long long* timing) {     // optional (5): per-phase cycle sums
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
    unsigned long long tacc[5] = {0, 0, 0, 0, 0};
#define PH_T0 unsigned long long tp0 = (timing && tid == 0) \
        ? __builtin_amdgcn_s_memtime() : 0
#define PH_T1(i) if (timing && tid == 0) \
        tacc[i] += __builtin_amdgcn_s_memtime() - tp0
    for (int wp = 0; wp < W; wp += 2) {
        const int w = wp + g;
        // build Hot^T: zero, then scatter 200 ones
        { PH_T0;
        if (!(dbg & 1u)) {
        for (int e = tidg; e < 16 * KP_LD; e += 256)
            (&lds.hot_t[g][0][0])[e] = f2bf(0.f);
        __syncthreads();
        for (int r = tidg; r < R; r += 256)
            lds.hot_t[g][lds.win[r * W + w]][r] = f2bf(1.0f);
        }
        __syncthreads();
        PH_T1(0); }

        // G1: A = W1 · Hot — each wave owns m-tiles {wid4, wid4+4} and runs
        // their two k-chains INTERLEAVED on split accumulators with the
        // shared b-fragment loaded once per k-step: the original
        // one-accumulator loop serialized load->load->mfma at ~170 cycles
        // per tile (phase bisection, profiles/) because every MFMA waited
        // on the previous one's accumulator and on freshly-issued LDS reads
        { PH_T0;
        if (!(dbg & 2u)) {
            const int mt0 = wid4, mt1 = wid4 + 4;
            const bool has1 = mt1 < 7;
            f32x4 acc0 = {0.f, 0.f, 0.f, 0.f}, acc1 = {0.f, 0.f, 0.f, 0.f};
            bf16x8 b0 = lds_load_b_frag_t(&lds.hot_t[g][0][0], 0, 0, KP_LD);
            bf16x8 a0 = lds_load_a_frag(&lds.w1t[0][0], mt0 * 16, 0, KP_LD);
            bf16x8 a1 = lds_load_a_frag(&lds.w1t[0][0], mt1 * 16, 0, KP_LD);
#pragma unroll
            for (int kb = 0; kb < 7; ++kb) {
                bf16x8 bn, an0, an1;
                if (kb + 1 < 7) {  // prefetch next k-step while MFMAs run
                    bn = lds_load_b_frag_t(&lds.hot_t[g][0][0], 0,
                                           (kb + 1) * 32, KP_LD);
                    an0 = lds_load_a_frag(&lds.w1t[0][0], mt0 * 16,
                                          (kb + 1) * 32, KP_LD);
                    an1 = lds_load_a_frag(&lds.w1t[0][0], mt1 * 16,
                                          (kb + 1) * 32, KP_LD);
                }
                acc0 = mfma16x16x32(a0, b0, acc0);
                if (has1) acc1 = mfma16x16x32(a1, b0, acc1);
                b0 = bn; a0 = an0; a1 = an1;
            }
#pragma unroll
            for (int i = 0; i < 4; ++i)
                lds.a_lds[g][mt0 * 16 + lrow * 4 + i][lcol] = f2bf(acc0[i]);
            if (has1)
#pragma unroll
                for (int i = 0; i < 4; ++i)
                    lds.a_lds[g][mt1 * 16 + lrow * 4 + i][lcol] = f2bf(acc1[i]);
        }
        // zero the k-pad of a_lds once: cols 12..15 are written with
        // zero-valued products (Hot cols 12..15 are zero); cols 16..31 are
        // cleared here on the first iteration only
        if (wp == 0)
            for (int e = tidg; e < MP; e += 256)
#pragma unroll
                for (int k = 16; k < 32; ++k) lds.a_lds[g][e][k] = f2bf(0.f);
        __syncthreads();
        PH_T1(1); }

        // G2: t1 = relu(A · E + b1) — 7 m-tiles x 4 n-tiles over 4 waves
        { PH_T0;
        if (!(dbg & 4u)) {
            // 7 tiles per wave (tile = wid4 + s*4 is always < 28): load all
            // operands first, then issue the 7 independent MFMAs, then the
            // epilogues — straight-line so nothing waits on LDS latency
            bf16x8 av[7], bv[7];
            f32x4 accv[7];
#pragma unroll
            for (int s = 0; s < 7; ++s) {
                const int tile = wid4 + s * 4;
                av[s] = lds_load_a_frag(&lds.a_lds[g][0][0], (tile >> 2) * 16,
                                        0, 40);
                bv[s] = lds_load_b_frag_t(&lds.e_t[0][0], (tile & 3) * 16, 0,
                                          32);
            }
#pragma unroll
            for (int s = 0; s < 7; ++s)
                accv[s] = mfma16x16x32(av[s], bv[s], f32x4{0.f, 0.f, 0.f, 0.f});
#pragma unroll
            for (int s = 0; s < 7; ++s) {
                const int tile = wid4 + s * 4;
                const int mt = tile >> 2, nt = tile & 3;
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    const int f = mt * 16 + lrow * 4 + i;
                    const int e = nt * 16 + lcol;
                    float v = accv[s][i] + (f < F1 ? lds.b1s[f] : 0.f);
                    lds.t1_t[g][e][f] = f2bf(fmaxf(v, 0.f));
                }
            }
        }
        __syncthreads();
        PH_T1(2); }

        // G3: t2 = relu(W2 · t1 + b2) — 4 n-tiles, one per wave
        { PH_T0;
        if (!(dbg & 8u)) {
            const int nt = wid4;
            // all 8 operand fragments up front, two independent k-chains
            bf16x8 a[4], bb[4];
#pragma unroll
            for (int kb = 0; kb < 4; ++kb) {
                a[kb] = lds_load_a_frag(&lds.w2_lds[0][0], 0, kb * 32, 136);
                bb[kb] = lds_load_b_frag_t(&lds.t1_t[g][0][0], nt * 16,
                                           kb * 32, 136);
            }
            f32x4 acc0 = mfma16x16x32(a[0], bb[0], f32x4{0.f, 0.f, 0.f, 0.f});
            f32x4 acc1 = mfma16x16x32(a[1], bb[1], f32x4{0.f, 0.f, 0.f, 0.f});
            acc0 = mfma16x16x32(a[2], bb[2], acc0);
            acc1 = mfma16x16x32(a[3], bb[3], acc1);
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int j = lrow * 4 + i;
                const int e = nt * 16 + lcol;
                if (j < F2 && e < E) {
                    float v = acc0[i] + acc1[i] + lds.b2s[j];
                    lds.t2[g][e * F2 + j] = f2bf(fmaxf(v, 0.f));
                }
            }
        }
        __syncthreads();
        PH_T1(3); }

        // coalesced store: out[(w, b), :] = t2 flattened (e*10+j)
        { PH_T0;
        if (!(dbg & 16u)) {
            bf16* dst = out + ((size_t)w * B + b) * OUT;
            if (tidg < 62)
                *reinterpret_cast<bf16x8*>(dst + tidg * 8) =
                    *reinterpret_cast<const bf16x8*>(&lds.t2[g][tidg * 8]);
            else if (tidg == 62)
#pragma unroll
                for (int q = 0; q < 4; ++q) dst[496 + q] = lds.t2[g][496 + q];
        }
        __syncthreads();
        PH_T1(4); }
    }
    if (timing && tid == 0)
#pragma unroll
        for (int i = 0; i < 5; ++i)
            atomicAdd(&timing[i], tacc[i]);
#undef PH_T0
#undef PH_T1
}

void embed_mlp_fwd(const uint8_t* ids, const void* w1, const float* b1,
                   const void* w2, const float* b2, const void* emb, void* out,
                   int B, hipStream_t stream, uint32_t dbg,
                   unsigned long long* timing) {
    hipLaunchKernelGGL(embed_mlp_fwd_kernel, dim3(B), dim3(512), 0, stream,
                       ids, static_cast<const bf16*>(w1), b1,
                       static_cast<const bf16*>(w2), b2,
                       static_cast<const bf16*>(emb), static_cast<bf16*>(out), B,
                       dbg, timing);
}

// ---------------------------------------------------------------------------
// v2: chunked-phase variant. The per-column version above runs 93 tiny MFMA
// tiles per column between 6 barriers and measures 49% issue-stall / 38%
// parked / 13% active (profiles/pmc_em, phase bisection in PERF_HISTORY.md).
// This variant processes FOUR columns per iteration as three WIDE GEMM
// phases over all 8 waves, so each phase is a long stream of independent
// tiles (fewer barriers per column, deeper load/MFMA pipelining, W1 read
// straight from L2):
//   P1: A4   (112 x 48)  = W1g (112x232_L2) · Hot4 (232 x 4*12)   147 MFMA
//   P2: t1_4 (4*112 x 64)= relu(A4 · E + b1)  per-column stacked  112 MFMA
//   P3: out  (16 x 208)  = relu(W2 · t1_4 + b2) -> global          91 MFMA
// Same math, same output, eval-exact.
namespace v2 {

constexpr int NC = 4;             // columns per chunk
constexpr int NCH = (W + NC - 1) / NC;  // 23 chunks (last partial, masked)
constexpr int N1 = NC * 12;       // G1 N = 48 (exactly 3 n-tiles)
constexpr int LD1 = 136;          // k-pad for the G3 operands (112 -> 136)

__global__ __launch_bounds__(512, 2) void embed_mlp_fwd2_kernel(
    const uint8_t* __restrict__ ids,  // (B, R, W)
    const bf16* __restrict__ w1g,     // (MP=112, KP_LD=232) zero-padded W1
    const float* __restrict__ b1,     // (F1)
    const bf16* __restrict__ w2,      // (F2, F1)
    const float* __restrict__ b2,     // (F2)
    const bf16* __restrict__ emb,     // (12, E)
    bf16* __restrict__ out,           // (W, B, OUT)
    int B) {
    __shared__ struct {
        uint8_t win[R * W];          // the window (staged once)
        bf16 hot_t[N1][KP_LD];       // G1 B-operand, [n = col*12+cls][k = r]
        bf16 a2[NC][MP][40];         // per-col G2 A-operand [f][c] (k-pad 40)
        bf16 e_t[64][32];            // G2 B-operand [e][c] (zero-padded)
        bf16 t1_t[208][LD1];         // G3 B-operand [n = col*50+e][k = f]
        bf16 w2_lds[16][LD1];        // G3 A-operand (zero-padded W2)
        float b1s[F1];
        float b2s[16];
    } lds;

    const int b = blockIdx.x;
    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int lrow = lane >> 4;
    const int lcol = lane & 15;

    // ---- one-time staging -------------------------------------------------
    // NOTE the full zero of a2 and t1_t: their k-pads (a2 cols 12-39, t1_t
    // cols 112-135) are multiplied by ZERO operands, which is only safe if
    // they are not NaN — uninitialized LDS can hold NaN bit patterns, and
    // 0 x NaN = NaN silently turns into 0 at the relu (fmaxf(NaN,0)=0),
    // zeroing whole output columns data-dependently. The valid regions are
    // rewritten every chunk; the pads stay zero for the whole kernel.
    for (int e = tid; e < 64 * 32; e += 512) (&lds.e_t[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < 16 * LD1; e += 512) (&lds.w2_lds[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < NC * MP * 40; e += 512)
        (&lds.a2[0][0][0])[e] = f2bf(0.f);
    for (int e = tid; e < 208 * LD1; e += 512)
        (&lds.t1_t[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < 16; e += 512) lds.b2s[e] = 0.f;
    __syncthreads();
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

    // hot is zeroed for chunk 0 here and for chunk ch+1 inside chunk ch's
    // P3 (different LDS arrays, so the stores overlap P3's MFMA streams and
    // the zero->scatter ordering rides P3's end barrier — one barrier saved
    // per chunk)
    auto zero_hot = [&]() {
        for (int e = tid * 8; e < N1 * KP_LD; e += 512 * 8)
            *reinterpret_cast<bf16x8*>(&lds.hot_t[0][0] + e) =
                bf16x8{f2bf(0.f), f2bf(0.f), f2bf(0.f), f2bf(0.f),
                       f2bf(0.f), f2bf(0.f), f2bf(0.f), f2bf(0.f)};
    };
    zero_hot();
    __syncthreads();

    for (int ch = 0; ch < NCH; ++ch) {
        const int c0 = ch * NC;
        // ---- P0: scatter 4 x 200 ones (hot zeroed by the previous chunk) --
        for (int idx = tid; idx < NC * R; idx += 512) {
            const int col = idx / R, r = idx - col * R;
            const int w = c0 + col;
            if (w < W)
                lds.hot_t[col * 12 + lds.win[r * W + w]][r] = f2bf(1.0f);
        }
        __syncthreads();

        // ---- P1: A4 = W1 · Hot4 — 21 tiles (7m x 3n) over 8 waves ---------
        for (int t = wid; t < 21; t += 8) {
            const int mt = t / 3, nt = t - mt * 3;
            bf16x8 av[7], bv[7];
#pragma unroll
            for (int kb = 0; kb < 7; ++kb) {
                av[kb] = global_load_a_frag(w1g, mt * 16, kb * 32, KP_LD);
                bv[kb] = lds_load_b_frag_t(&lds.hot_t[0][0], nt * 16, kb * 32,
                                           KP_LD);
            }
            f32x4 acc0 = {0.f, 0.f, 0.f, 0.f}, acc1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int kb = 0; kb < 7; kb += 2) acc0 = mfma16x16x32(av[kb], bv[kb], acc0);
#pragma unroll
            for (int kb = 1; kb < 7; kb += 2) acc1 = mfma16x16x32(av[kb], bv[kb], acc1);
            const int n = nt * 16 + lcol;
            const int col = n / 12, c = n - col * 12;
#pragma unroll
            for (int i = 0; i < 4; ++i)
                lds.a2[col][mt * 16 + lrow * 4 + i][c] = f2bf(acc0[i] + acc1[i]);
        }
        __syncthreads();

        // ---- P2: t1 = relu(A4 · E + b1) — 112 single-MFMA tiles -----------
        {
            bf16x8 av[14], bv[14];
            f32x4 accv[14];
#pragma unroll
            for (int s = 0; s < 14; ++s) {
                const int t = wid + s * 8;
                const int col = t / 28, rem = t - col * 28;
                const int mt = rem >> 2, nt = rem & 3;
                av[s] = lds_load_a_frag(&lds.a2[col][0][0], mt * 16, 0, 40);
                bv[s] = lds_load_b_frag_t(&lds.e_t[0][0], nt * 16, 0, 32);
            }
#pragma unroll
            for (int s = 0; s < 14; ++s)
                accv[s] = mfma16x16x32(av[s], bv[s], f32x4{0.f, 0.f, 0.f, 0.f});
#pragma unroll
            for (int s = 0; s < 14; ++s) {
                const int t = wid + s * 8;
                const int col = t / 28, rem = t - col * 28;
                const int mt = rem >> 2, nt = rem & 3;
                const int e = nt * 16 + lcol;
                if (e < E) {
#pragma unroll
                    for (int i = 0; i < 4; ++i) {
                        const int f = mt * 16 + lrow * 4 + i;
                        float v = accv[s][i] + (f < F1 ? lds.b1s[f] : 0.f);
                        lds.t1_t[col * E + e][f] = f2bf(fmaxf(v, 0.f));
                    }
                }
            }
        }
        __syncthreads();

        // ---- P3: out = relu(W2 · t1 + b2), stored straight to global ------
        if (ch + 1 < NCH) zero_hot();
        for (int t = wid; t < 13; t += 8) {
            bf16x8 av[4], bv[4];
#pragma unroll
            for (int kb = 0; kb < 4; ++kb) {
                av[kb] = lds_load_a_frag(&lds.w2_lds[0][0], 0, kb * 32, LD1);
                bv[kb] = lds_load_b_frag_t(&lds.t1_t[0][0], t * 16, kb * 32,
                                           LD1);
            }
            f32x4 acc0 = mfma16x16x32(av[0], bv[0], f32x4{0.f, 0.f, 0.f, 0.f});
            f32x4 acc1 = mfma16x16x32(av[1], bv[1], f32x4{0.f, 0.f, 0.f, 0.f});
            acc0 = mfma16x16x32(av[2], bv[2], acc0);
            acc1 = mfma16x16x32(av[3], bv[3], acc1);
            const int n = t * 16 + lcol;
            const int col = n / E, e = n - col * E;
            const int w = c0 + col;
            if (n < NC * E && w < W) {
                bf16* dst = out + ((size_t)w * B + b) * OUT + e * F2;
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    const int j = lrow * 4 + i;
                    if (j < F2) {
                        float v = acc0[i] + acc1[i] + lds.b2s[j];
                        dst[j] = f2bf(fmaxf(v, 0.f));
                    }
                }
            }
        }
        __syncthreads();
    }
}

}  // namespace v2

// ---------------------------------------------------------------------------
// v3: wave-private columns. v2's cost is NOT MFMA (the ~7.2k MFMAs per
// window are ~13 us of pipe time vs 156 us measured): its 4 barriers x 23
// chunks park every wave on the slowest phase and the staged tiles fight
// over LDS banks (PMC: ~50% parked / 43% conflicts). Here each WAVE owns a
// whole column with PRIVATE hot/a/t1/t2 tiles — there is no barrier between
// G1/G2/G3 at all (within-wave lgkmcnt ordering is free), so a wave streams
// 93 MFMAs per column back to back with register prefetch across phases.
// 4 waves per workgroup (LDS: 4 x ~35 KB private + ~9 KB shared = 148 KB);
// W1 A-fragments come straight from L2 like v2's P1. One wave per SIMD —
// the stream is MFMA-ILP-rich, so no partner wave is needed for cover.
namespace v3 {

constexpr int WAVES3 = 4;
constexpr int LD1 = 136;  // t1_t/w2 row stride (k-pad 112 -> 136)

__global__ __launch_bounds__(WAVES3 * 64, 1) void embed_mlp_fwd3_kernel(
    const uint8_t* __restrict__ ids,  // (B, R, W)
    const bf16* __restrict__ w1g,     // (MP=112, KP_LD=232) zero-padded W1
    const float* __restrict__ b1,     // (F1)
    const bf16* __restrict__ w2,      // (F2, F1)
    const float* __restrict__ b2,     // (F2)
    const bf16* __restrict__ emb,     // (12, E)
    bf16* __restrict__ out,           // (W, B, OUT)
    int B,
    unsigned long long* __restrict__ timing) {  // optional (6): per-phase
                                                // cycle sums from lane 0
    __shared__ struct {
        // shared, read-only after staging
        bf16 e_t[64][32];        // G2 B-operand [e][k=c] (zero-padded)
        bf16 w2_lds[16][LD1];    // G3 A-operand (zero-padded W2)
        float b1s[F1];
        float b2s[16];
        // per-wave private tiles
        bf16 hot_t[WAVES3][16][KP_LD];  // G1 B-operand [cls][k=r]
        bf16 a[WAVES3][MP][40];         // G1 out / G2 A-operand [f][c]
        bf16 t1_t[WAVES3][64][LD1];     // G3 B-operand [e][k=f]
        bf16 t2[WAVES3][512];           // output staging
    } lds;

    const int b = blockIdx.x;
    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int lrow = lane >> 4;
    const int lcol = lane & 15;

    // ---- one-time staging (zero pads once; see v2's NaN note) -------------
    for (int e = tid; e < 64 * 32; e += WAVES3 * 64) (&lds.e_t[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < 16 * LD1; e += WAVES3 * 64)
        (&lds.w2_lds[0][0])[e] = f2bf(0.f);
    for (int e = tid; e < WAVES3 * MP * 40; e += WAVES3 * 64)
        (&lds.a[0][0][0])[e] = f2bf(0.f);
    for (int e = tid; e < WAVES3 * 64 * LD1; e += WAVES3 * 64)
        (&lds.t1_t[0][0][0])[e] = f2bf(0.f);
    for (int e = tid; e < 16; e += WAVES3 * 64) lds.b2s[e] = 0.f;
    __syncthreads();
    for (int e = tid; e < 12 * E; e += WAVES3 * 64) lds.e_t[e % E][e / E] = emb[e];
    for (int e = tid; e < F2 * F1; e += WAVES3 * 64)
        lds.w2_lds[e / F1][e % F1] = w2[e];
    for (int e = tid; e < F1; e += WAVES3 * 64) lds.b1s[e] = b1[e];
    for (int e = tid; e < F2; e += WAVES3 * 64) lds.b2s[e] = b2[e];
    // private hot tiles: zero whole (incl. pad rows 12-15) once; the scatter
    // epilogue below re-zeroes exactly the 200 entries it set
    for (int e = tid; e < WAVES3 * 16 * KP_LD; e += WAVES3 * 64)
        (&lds.hot_t[0][0][0])[e] = f2bf(0.f);
    __syncthreads();  // the ONLY barrier (staging); columns are wave-private

    bf16(&hot)[16][KP_LD] = lds.hot_t[wid];
    bf16(&at)[MP][40] = lds.a[wid];
    bf16(&t1)[64][LD1] = lds.t1_t[wid];
    bf16* t2 = lds.t2[wid];
    const uint8_t* win = ids + (size_t)b * R * W;

    // loop-invariant operand fragments + biases, register-resident for all
    // columns (the per-column scalar b1s/b2s LDS reads were ~112/col)
    bf16x8 e_bv[4], w2_av[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt)
        e_bv[nt] = lds_load_b_frag_t(&lds.e_t[0][0], nt * 16, 0, 32);
#pragma unroll
    for (int kb = 0; kb < 4; ++kb)
        w2_av[kb] = lds_load_a_frag(&lds.w2_lds[0][0], 0, kb * 32, LD1);
    float b1r[7][4], b2r[4];
#pragma unroll
    for (int mt = 0; mt < 7; ++mt)
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int f = mt * 16 + lrow * 4 + i;
            b1r[mt][i] = (f < F1) ? lds.b1s[f] : 0.f;
        }
#pragma unroll
    for (int i = 0; i < 4; ++i)
        b2r[i] = lds.b2s[lrow * 4 + i];

    unsigned long long tacc[6] = {0, 0, 0, 0, 0, 0};
#define V3_T0 unsigned long long tp0 = (timing && tid == 0) \
        ? __builtin_amdgcn_s_memtime() : 0
#define V3_T1(i) if (timing && tid == 0) { \
        unsigned long long tn = __builtin_amdgcn_s_memtime(); \
        tacc[i] += tn - tp0; tp0 = tn; }
    // wave w handles columns w, w+WAVES3, ... — each fully privately
    for (int w = wid; w < W; w += WAVES3) {
        V3_T0;
        // ---- scatter: 200 one-hots (ids read straight through L1) ---------
        // lane r and r+64... handle reads r, r+64, r+128 (200 = 3*64 + 8)
        uint8_t myid[4];
        int nr = 0;
#pragma unroll
        for (int q = 0; q < 4; ++q) {
            const int r = lane + q * 64;
            if (r < R) myid[nr++] = win[(size_t)r * W + w];
        }
#pragma unroll
        for (int q = 0; q < 4; ++q)
            if (q < nr) hot[myid[q]][lane + q * 64] = f2bf(1.0f);
        V3_T1(0);

        // ---- G1: A = W1 · Hot  (7 m-tiles x 7 k-steps, 49 MFMA) -----------
        // 7 independent accumulator chains; B-frag per k shared across m.
        // THREE-deep k-window: at one wave per SIMD the register budget is
        // 512/lane, so ~21 L2 A-fragments stay in flight — ~2 k-steps
        // (~240 MFMA cycles) of cover for the ~200-cycle L2 latency (the
        // 1-deep window measured neutral: 7 MFMAs of cover was not enough).
        {
            f32x4 acc[7];
#pragma unroll
            for (int mt = 0; mt < 7; ++mt) acc[mt] = f32x4{0.f, 0.f, 0.f, 0.f};
            bf16x8 af[3][7];
            bf16x8 bf3[3];
#pragma unroll
            for (int p = 0; p < 2; ++p) {
#pragma unroll
                for (int mt = 0; mt < 7; ++mt)
                    af[p][mt] = global_load_a_frag(w1g, mt * 16, p * 32,
                                                   KP_LD);
                bf3[p] = lds_load_b_frag_t(&hot[0][0], 0, p * 32, KP_LD);
            }
#pragma unroll
            for (int kb = 0; kb < 7; ++kb) {
                const int cur = kb % 3, nxt = (kb + 2) % 3;
                if (kb + 2 < 7) {
#pragma unroll
                    for (int mt = 0; mt < 7; ++mt)
                        af[nxt][mt] = global_load_a_frag(w1g, mt * 16,
                                                         (kb + 2) * 32,
                                                         KP_LD);
                    bf3[nxt] = lds_load_b_frag_t(&hot[0][0], 0,
                                                 (kb + 2) * 32, KP_LD);
                }
#pragma unroll
                for (int mt = 0; mt < 7; ++mt)
                    acc[mt] = mfma16x16x32(af[cur][mt], bf3[cur], acc[mt]);
            }
            // epilogue: a[f][c] (c = lcol < 12 live, 12..15 zero products)
#pragma unroll
            for (int mt = 0; mt < 7; ++mt)
#pragma unroll
                for (int i = 0; i < 4; ++i)
                    at[mt * 16 + lrow * 4 + i][lcol] = f2bf(acc[mt][i]);
            // un-scatter the hot ones (cheaper than re-zeroing 7.4 KB)
#pragma unroll
            for (int q = 0; q < 4; ++q)
                if (q < nr) hot[myid[q]][lane + q * 64] = f2bf(0.0f);
        }
        V3_T1(1);

        // ---- G2: t1 = relu(A · E + b1)  (7m x 4n single-K, 28 MFMA) -------
        {
            f32x4 acc[7][4];
            bf16x8 av[7];
#pragma unroll
            for (int mt = 0; mt < 7; ++mt)
                av[mt] = lds_load_a_frag(&at[0][0], mt * 16, 0, 40);
#pragma unroll
            for (int mt = 0; mt < 7; ++mt)
#pragma unroll
                for (int nt = 0; nt < 4; ++nt)
                    acc[mt][nt] = mfma16x16x32(av[mt], e_bv[nt],
                                               f32x4{0.f, 0.f, 0.f, 0.f});
#pragma unroll
            for (int mt = 0; mt < 7; ++mt)
#pragma unroll
                for (int nt = 0; nt < 4; ++nt) {
                    const int e = nt * 16 + lcol;
                    if (e < E) {
                        // f = mt*16 + lrow*4 + i contiguous over i: one b64
                        bf16x4 pk;
#pragma unroll
                        for (int i = 0; i < 4; ++i) {
                            const float v = acc[mt][nt][i] + b1r[mt][i];
                            pk[i] = f2bf(fmaxf(v, 0.f));
                        }
                        *reinterpret_cast<bf16x4*>(
                            &t1[e][mt * 16 + lrow * 4]) = pk;
                    }
                }
        }

        V3_T1(2);
        // ---- G3: t2 = relu(W2 · t1 + b2)  (4 n-tiles x K=112, 16 MFMA) ----
        {
#pragma unroll
            for (int nt = 0; nt < 4; ++nt) {
                f32x4 acc0 = {0.f, 0.f, 0.f, 0.f}, acc1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
                for (int kb = 0; kb < 4; kb += 2) {
                    acc0 = mfma16x16x32(
                        w2_av[kb],
                        lds_load_b_frag_t(&t1[0][0], nt * 16, kb * 32, LD1),
                        acc0);
                    acc1 = mfma16x16x32(
                        w2_av[kb + 1],
                        lds_load_b_frag_t(&t1[0][0], nt * 16, (kb + 1) * 32,
                                          LD1),
                        acc1);
                }
                const int e = nt * 16 + lcol;
                const int j = lrow * 4;  // rows j..j+3; only j<10 live
                if (e < E) {
#pragma unroll
                    for (int i = 0; i < 4; ++i)
                        if (j + i < F2) {
                            const float v = acc0[i] + acc1[i] + b2r[i];
                            t2[e * F2 + j + i] = f2bf(fmaxf(v, 0.f));
                        }
                }
            }
        }

        V3_T1(3);
        // ---- coalesced store (wave-wide b128) -----------------------------
        {
            bf16* dst = out + ((size_t)w * B + b) * OUT;
            const int e8 = lane * 8;
            if (e8 + 8 <= OUT)
                *reinterpret_cast<bf16x8*>(dst + e8) =
                    *reinterpret_cast<const bf16x8*>(&t2[e8]);
            else if (e8 < OUT)
                for (int q = e8; q < OUT; ++q) dst[q] = t2[q];
        }
        V3_T1(4);
    }
    if (timing && tid == 0)
#pragma unroll
        for (int i = 0; i < 5; ++i) atomicAdd(&timing[i], tacc[i]);
#undef V3_T0
#undef V3_T1
}

}  // namespace v3

void embed_mlp_fwd3(const uint8_t* ids, const void* w1g, const float* b1,
                    const void* w2, const float* b2, const void* emb,
                    void* out, int B, hipStream_t stream,
                    unsigned long long* timing) {
    hipLaunchKernelGGL(v3::embed_mlp_fwd3_kernel, dim3(B),
                       dim3(v3::WAVES3 * 64), 0, stream, ids,
                       static_cast<const bf16*>(w1g), b1,
                       static_cast<const bf16*>(w2), b2,
                       static_cast<const bf16*>(emb), static_cast<bf16*>(out),
                       B, timing);
}

void embed_mlp_fwd2(const uint8_t* ids, const void* w1g, const float* b1,
                    const void* w2, const float* b2, const void* emb,
                    void* out, int B, hipStream_t stream) {
    hipLaunchKernelGGL(v2::embed_mlp_fwd2_kernel, dim3(B), dim3(512), 0,
                       stream, ids, static_cast<const bf16*>(w1g), b1,
                       static_cast<const bf16*>(w2), b2,
                       static_cast<const bf16*>(emb), static_cast<bf16*>(out),
                       B);
}

}  // namespace rk
