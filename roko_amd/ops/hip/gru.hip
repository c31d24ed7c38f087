// Persistent bidirectional GRU layer forward + BPTT backward for gfx950.
//
// Replaces the reference's cuDNN GRU (rnn_model.py:57).
// Design (SURVEY.md §7 hard part (a), re-thought for CDNA4):
//   * the input projections W_ih·x + b_ih for ALL T steps are one plain GEMM
//     done on the host side (hipBLASLt via torch.matmul) — this kernel gets
//     the precomputed per-step gate inputs `xg`;
//   * ONE kernel launch runs the full T-step recurrence of a layer: grid =
//     (B/MBT batch tiles) x (2 directions); each workgroup owns MBT batch
//     rows (default 16 — the gate phase is VALU-issue-floor bound, so the
//     smallest MFMA-viable tile halves the per-step wall AND doubles the
//     workgroup supply for serving packing; docs/KERNELS.md lesson 18)
//     whose hidden state lives in fp32 registers with a bf16 mirror in LDS
//     as the MFMA A-operand — no per-step launches, no grid-wide sync;
//   * 8 waves per workgroup = two per SIMD: the matrix and VALU pipes are
//     separate and arbitrate between co-resident waves, so one wave's
//     dependent gate math overlaps its partner's MFMAs;
//   * U = weight_hh^T fragments are loaded ONCE into registers (12 bf16x8
//     per wave) and reused for all T steps; per-step work is 12 (MBT=16)
//     or 24 (MBT=32) v_mfma_f32_16x16x32_bf16 per wave + fused
//     sigmoid/tanh gate math;
//   * the hidden-state mirror is double-buffered in LDS, so each step needs
//     exactly ONE __syncthreads(): MFMAs read h[cur] while gate math writes
//     h[cur^1]; the barrier publishes h[cur^1] for the next step;
//   * per-step xg tiles (32x384 bf16) are double-buffered through LDS with a
//     prefetch distance of TWO steps (issue-early / write-late, T14): the
//     loads for step t+2 issue before anything waits on step t+1's, hiding
//     the full HBM/L3 latency under one whole step of compute.
//
// PyTorch GRU semantics (gate order r,z,n in weight rows; n-gate bias split):
//   r = sigmoid(xr + U_r h + bhh_r)
//   z = sigmoid(xz + U_z h + bhh_z)
//   n = tanh(xn + r * (U_n h + bhh_n))        (xg already contains b_ih)
//   h' = (1 - z) * n + z * h

#include <cstdint>
#include <cstdlib>
#include <string>
#include <type_traits>

#include "common.h"

namespace rk {

constexpr int H = 128;       // hidden size (config.HIDDEN_SIZE)
constexpr int G3 = 3 * H;    // gate rows
constexpr int MB = 32;       // batch rows per workgroup
constexpr int WAVES = 8;     // 512 threads, two waves per SIMD: a wave's
                             // dependent gate VALU hides under its partner's
                             // MFMAs (MI355X_MICROARCH.md "Two waves per
                             // SIMD" — the pipes are separate and arbitrate)
constexpr int HPAD = H + 8;  // LDS row padding (bank-conflict fix)
constexpr int XCH = MB * G3 / (WAVES * 64 * 8);  // xg chunks per thread (3)

constexpr int XKP = 512;        // max padded input width for the fused xg GEMM
constexpr int XLD = XKP + 8;    // x-stage LDS row stride (bank pad)

// FUSEXG (serving variant): the input-projection GEMM xg = x·W_ihᵀ + b_ih
// runs as PHASE 1 of this kernel instead of a separate hipBLASLt launch.
// Rationale (profiles/infer_r02_v3_kernel_stats.csv): hipBLASLt runs the
// (11520, 768, K<=500) GEMM at ~543 TF/s on ~225 workgroups — ~105 CU·us
// per window of aggregate chip time, tied with the front as the top serving
// cost. Here each of the 8 workgroups computes its OWN (90x32, 384) slice
// with the W slice L2-resident and reused across all 90 steps, at a small
// fraction of the aggregate CU time; the per-chain latency rises (~+60 us),
// which deep pipelining absorbs. Not used by training: there the step's
// LATENCY is the metric and 225 workgroups beat 8.
template <bool TRAIN, int FUSEKP = 0, int MBT = MB>
// FUSEKP: 0 = xg precomputed; 256/512 = in-kernel xg GEMM width.
// MBT: batch rows per workgroup. 32 is the default; 16 (eval only) doubles
// the workgroup count and HALVES each thread's gate-chain count — the gate
// phase sits at the 2-wave VALU-issue floor, so per-step time drops with
// per-thread instruction count, and b=128 serving gets 16 WGs of supply.
__global__ __launch_bounds__(WAVES * 64, 2) void gru_layer_fwd_kernel(
    const bf16* __restrict__ xg,   // (T, B, 2, 3H)  W_ih·x + b_ih
                                   // (FUSEXG: workspace this kernel fills)
    const bf16* __restrict__ u,    // (2, 3H, H)     weight_hh
    const float* __restrict__ bhh, // (2, 3H)        bias_hh
    bf16* __restrict__ hseq,       // (T, B, 2, H)   output
    bf16* __restrict__ cache,      // (T, B, 2, H, 4) [r z n hgn] or nullptr
    int T, int B,
    uint32_t dbg,   // timing bisection: 1 no hseq store, 2 no cache store,
                    // 4 no gate VALU, 8 no xg staging, 16 no MFMA
    const bf16* __restrict__ x = nullptr,     // (T, B, IN) layer input
    const bf16* __restrict__ w_ih_p = nullptr,  // (768, KP) row-padded W_ih
    const bf16* __restrict__ b_ih = nullptr,   // (768)
    int IN = 0) {
    constexpr bool FUSEXG = FUSEKP > 0;
    static_assert(!FUSEXG || MBT == 32, "fused xg path is MB=32 only");
    constexpr int MT = MBT / 16;         // 16-row m-subtiles per thread
    constexpr int CW = (MBT == 32) ? 8 : 4;  // staging chunk width (bf16)
    constexpr int XCHT = MBT * G3 / (WAVES * 64 * CW);  // xg chunks/thread
    using xvec = std::conditional_t<CW == 8, bf16x8, bf16x4>;
    constexpr int XLDK = FUSEXG ? FUSEKP + 8 : 1;
    __shared__ struct {
        bf16 h[2][MBT][HPAD];        // double-buffered hidden-state mirror
        bf16 xgb[2][MBT][G3];        // double-buffered step gate inputs
        bf16 x_st[FUSEXG ? 2 : 1][FUSEXG ? MBT : 1][XLDK];
    } lds;

    const int dir = blockIdx.y;
    const int b0 = blockIdx.x * MBT;
    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int j0 = wid * 16;         // this wave's 16 hidden columns
    const int lrow = lane >> 4;      // fragment row group (0..3)
    const int lcol = lane & 15;      // fragment column
    // static priority for the second-dispatched wave half: the younger
    // wave on each SIMD loses VALU arbitration at every barrier-paced
    // segment (MI355X_MICROARCH.md "Two waves per SIMD" item 4)
    if (wid >= WAVES / 2) __builtin_amdgcn_s_setprio(1);

    // ---- PHASE 1 (FUSEXG): xg[t, b0:b0+MB, dir, :] = x·Wᵀ + b_ih ----------
    if constexpr (FUSEXG) {
        // x[t] tile (MB, KP) double-buffered through LDS; W B-fragments read
        // straight from L2 ((768, KP) row-major = [col][k], so the B-frag
        // load is the A-frag address pattern). Wave owns 3 n-tiles
        // (nt = wid + 8s) x 2 m-tiles; one barrier per t. KP is a template
        // constant: the k-loop fully unrolls and the W/A fragment loads for
        // kb+1 issue under kb's MFMAs (a runtime trip count compiled to a
        // serial load->mfma->load chain — 1.47 ms/layer vs ~210 split).
        constexpr int KB = FUSEKP / 32;
        auto stage_x = [&](int t, int buf) {
            const bf16* src = x + ((size_t)t * B + b0) * IN;
#pragma unroll
            for (int p = 0; p < MBT * FUSEKP / 4 / (WAVES * 64); ++p) {
                const int c = p * WAVES * 64 + tid;
                const int row = (c * 4) / FUSEKP, col = (c * 4) % FUSEKP;
                uint64_t v = 0;  // 4 bf16 (zero pads cols IN..KP)
                if (col + 4 <= IN)
                    v = *reinterpret_cast<const uint64_t*>(
                        src + (size_t)row * IN + col);
                *reinterpret_cast<uint64_t*>(&lds.x_st[buf][row][col]) = v;
            }
        };
        float bih_reg[3];
#pragma unroll
        for (int s = 0; s < 3; ++s) {
            const int col = (wid + 8 * s) * 16 + lcol;
            bih_reg[s] = b_ih ? bf2f(b_ih[dir * G3 + col]) : 0.f;
        }
        const int col0w = dir * G3 + wid * 16;
        stage_x(0, 0);
        __syncthreads();
        for (int t = 0; t < T; ++t) {
            if (t + 1 < T) stage_x(t + 1, (t + 1) & 1);
            f32x4 acc[3][2];
#pragma unroll
            for (int s = 0; s < 3; ++s)
#pragma unroll
                for (int mt = 0; mt < 2; ++mt)
                    acc[s][mt] = f32x4{0.f, 0.f, 0.f, 0.f};
            bf16x8 wfA[3], wfB[3], a0A, a1A, a0B, a1B;
#pragma unroll
            for (int s = 0; s < 3; ++s)
                wfA[s] = global_load_a_frag(w_ih_p, col0w + s * 128, 0,
                                            FUSEKP);
            a0A = lds_load_a_frag(&lds.x_st[t & 1][0][0], 0, 0, XLDK);
            a1A = lds_load_a_frag(&lds.x_st[t & 1][0][0], 16, 0, XLDK);
#pragma unroll
            for (int kb = 0; kb < KB; ++kb) {
                bf16x8(&wc)[3] = (kb & 1) ? wfB : wfA;
                bf16x8(&wn)[3] = (kb & 1) ? wfA : wfB;
                const bf16x8 a0 = (kb & 1) ? a0B : a0A;
                const bf16x8 a1 = (kb & 1) ? a1B : a1A;
                if (kb + 1 < KB) {
#pragma unroll
                    for (int s = 0; s < 3; ++s)
                        wn[s] = global_load_a_frag(w_ih_p, col0w + s * 128,
                                                   (kb + 1) * 32, FUSEKP);
                    ((kb & 1) ? a0A : a0B) = lds_load_a_frag(
                        &lds.x_st[t & 1][0][0], 0, (kb + 1) * 32, XLDK);
                    ((kb & 1) ? a1A : a1B) = lds_load_a_frag(
                        &lds.x_st[t & 1][0][0], 16, (kb + 1) * 32, XLDK);
                }
#pragma unroll
                for (int s = 0; s < 3; ++s) {
                    acc[s][0] = mfma16x16x32(a0, wc[s], acc[s][0]);
                    acc[s][1] = mfma16x16x32(a1, wc[s], acc[s][1]);
                }
            }
            // epilogue: + b_ih, scalar bf16 stores into the xg workspace
            // (xg is a mutable workspace in the FUSEXG instantiation)
            bf16* dst = const_cast<bf16*>(xg) +
                        (((size_t)t * B + b0) * 2 + dir) * G3;
#pragma unroll
            for (int s = 0; s < 3; ++s) {
                const int col = (wid + 8 * s) * 16 + lcol;
#pragma unroll
                for (int mt = 0; mt < 2; ++mt)
#pragma unroll
                    for (int i = 0; i < 4; ++i) {
                        const int row = mt * 16 + lrow * 4 + i;
                        dst[(size_t)row * 2 * G3 + col] =
                            f2bf(acc[s][mt][i] + bih_reg[s]);
                    }
            }
            __syncthreads();
        }
        // all xg stores must land before phase 2 reads them back
        __builtin_amdgcn_s_waitcnt(0);  // vmcnt(0) & lgkmcnt(0)
        __syncthreads();
    }

    // ---- load U fragments (kept in registers for all T steps) -------------
    // B-fragment for gates = h·U^T: B[k][col] = U[gate*H + j0 + 16ct + col][k]
    bf16x8 ufrag[3][4];
#pragma unroll
    for (int g = 0; g < 3; ++g)
#pragma unroll
        for (int kb = 0; kb < 4; ++kb) {
            const int jrow = g * H + j0 + lcol;
            const int k = kb * 32 + 8 * lrow;
            ufrag[g][kb] = *reinterpret_cast<const bf16x8*>(
                u + (size_t)dir * G3 * H + (size_t)jrow * H + k);
        }
    float bhh_reg[3];
#pragma unroll
    for (int g = 0; g < 3; ++g) bhh_reg[g] = bhh[dir * G3 + g * H + j0 + lcol];

    // ---- zero hidden state -------------------------------------------------
    float hreg[MT][4];  // [mt][i] fp32 master copy, fragment-shaped
#pragma unroll
    for (int mt = 0; mt < MT; ++mt)
#pragma unroll
        for (int i = 0; i < 4; ++i) hreg[mt][i] = 0.0f;
    for (int e = tid; e < MBT * HPAD; e += WAVES * 64) lds.h[0][0][e] = f2bf(0.0f);

    const int stp = (dir == 0) ? 1 : -1;
    const int tfirst = (dir == 0) ? 0 : T - 1;
    auto xg_src = [&](int t) {
        return xg + (((size_t)t * B + b0) * 2 + dir) * G3;
    };

    // ---- prologue: xg[t0] -> LDS buffer 0; xg[t1] -> stage registers ------
    {
        const bf16* src = xg_src(tfirst);
#pragma unroll
        for (int p = 0; p < XCHT; ++p) {
            int e = (p * WAVES * 64 + tid) * CW;
            int row = e / G3, col = e % G3;
            *reinterpret_cast<xvec*>(&lds.xgb[0][row][col]) =
                *reinterpret_cast<const xvec*>(src + (size_t)row * 2 * G3 + col);
        }
    }
    xvec stageA[XCHT], stageB[XCHT];
    if (T > 1) {
        const bf16* src = xg_src(tfirst + stp);
#pragma unroll
        for (int p = 0; p < XCHT; ++p) {
            int e = (p * WAVES * 64 + tid) * CW;
            int row = e / G3, col = e % G3;
            stageA[p] =
                *reinterpret_cast<const xvec*>(src + (size_t)row * 2 * G3 + col);
        }
    }
    __syncthreads();

    // ---- T-step recurrence, ONE barrier per step --------------------------
    // The loop is unrolled by TWO with alternating stage register sets so no
    // register copy ever waits on the loads issued in the SAME step: the xg
    // write into LDS only touches registers loaded a FULL step earlier (the
    // v3 kernel's stage rotation copy forced an end-of-step vmcnt(0) on
    // loads ~200 cycles old — profiles/pmc_gru_r01.txt's 36% SQ_WAIT_ANY).
    auto body = [&](int ti, int curp, xvec (&st_wr)[XCHT],
                    xvec (&st_ld)[XCHT]) {
        const int t = tfirst + stp * ti;
        // issue xg[t+2] loads FIRST (they retire next step), then write the
        // staged xg[t+1] (loaded last step) into the back LDS buffer
        if (ti + 2 < T && !(dbg & 8u)) {
            const bf16* src = xg_src(t + 2 * stp);
#pragma unroll
            for (int p = 0; p < XCHT; ++p) {
                int e = (p * WAVES * 64 + tid) * CW;
                int row = e / G3, col = e % G3;
                st_ld[p] =
                    *reinterpret_cast<const xvec*>(src + (size_t)row * 2 * G3 + col);
            }
        }
        if (ti + 1 < T && !(dbg & 8u)) {
#pragma unroll
            for (int p = 0; p < XCHT; ++p) {
                int e = (p * WAVES * 64 + tid) * CW;
                int row = e / G3, col = e % G3;
                *reinterpret_cast<xvec*>(&lds.xgb[curp ^ 1][row][col]) = st_wr[p];
            }
        }

        // gates_h = h · U^T  (24 MFMA per wave, A-frags shared 3-ways)
        f32x4 acc[MT][3];
#pragma unroll
        for (int mt = 0; mt < MT; ++mt)
#pragma unroll
            for (int g = 0; g < 3; ++g) acc[mt][g] = f32x4{0.f, 0.f, 0.f, 0.f};
        if (!(dbg & 16u)) {
#pragma unroll
            for (int kb = 0; kb < 4; ++kb) {
#pragma unroll
                for (int mt = 0; mt < MT; ++mt) {
                    bf16x8 a = lds_load_a_frag(&lds.h[curp][0][0], mt * 16, kb * 32, HPAD);
#pragma unroll
                    for (int g = 0; g < 3; ++g)
                        acc[mt][g] = mfma16x16x32(a, ufrag[g][kb], acc[mt][g]);
                }
            }
        }

        // fused gate math; updates the fp32 register hidden state and writes
        // the bf16 mirror into the BACK h buffer (no reader until barrier).
        // Structured as: batch ALL 24 xg reads first, then run the 8
        // elements' serial exp/rcp chains as INDEPENDENT interleavable
        // streams — the original per-element form (reads + chain + a dbg
        // branch per element) compiled to 16 fenced basic blocks whose
        // ~200-cycle dependency chains executed back to back: the gate
        // phase alone was 55% of the kernel (scripts/gru_timing.py).
        if (dbg & 4u) {  // timing: gate math stripped (separate cold loop)
#pragma unroll
            for (int mt = 0; mt < MT; ++mt)
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    const int row = mt * 16 + lrow * 4 + i;
                    const int j = j0 + lcol;
                    const float hnew =
                        acc[mt][0][i] + acc[mt][1][i] + acc[mt][2][i];
                    hreg[mt][i] = hnew;
                    lds.h[curp ^ 1][row][j] = f2bf(hnew);
                    if constexpr (TRAIN) {
                        if (!(dbg & 2u)) {
                            bf16x4 pk = {f2bf(hnew), f2bf(hnew), f2bf(hnew),
                                         f2bf(hnew)};
                            *reinterpret_cast<bf16x4*>(
                                cache +
                                (((size_t)t * B + b0 + row) * 2 + dir) * 4 * H +
                                4 * j) = pk;
                        }
                    }
                }
        } else {
            const int j = j0 + lcol;
            float xr[MT][4], xz[MT][4], xn[MT][4];
#pragma unroll
            for (int mt = 0; mt < MT; ++mt)
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    const int row = mt * 16 + lrow * 4 + i;
                    xr[mt][i] = bf2f(lds.xgb[curp][row][0 * H + j]);
                    xz[mt][i] = bf2f(lds.xgb[curp][row][1 * H + j]);
                    xn[mt][i] = bf2f(lds.xgb[curp][row][2 * H + j]);
                }
            float r8[MT][4], z8[MT][4], n8[MT][4], hg8[MT][4];
#pragma unroll
            for (int mt = 0; mt < MT; ++mt)
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    hg8[mt][i] = acc[mt][2][i] + bhh_reg[2];
                    r8[mt][i] =
                        sigmoidf_dev(xr[mt][i] + acc[mt][0][i] + bhh_reg[0]);
                    z8[mt][i] =
                        sigmoidf_dev(xz[mt][i] + acc[mt][1][i] + bhh_reg[1]);
                }
#pragma unroll
            for (int mt = 0; mt < MT; ++mt)
#pragma unroll
                for (int i = 0; i < 4; ++i)
                    n8[mt][i] = tanhf_dev(xn[mt][i] + r8[mt][i] * hg8[mt][i]);
#pragma unroll
            for (int mt = 0; mt < MT; ++mt)
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    const int row = mt * 16 + lrow * 4 + i;
                    const float hnew = (1.0f - z8[mt][i]) * n8[mt][i] +
                                       z8[mt][i] * hreg[mt][i];
                    hreg[mt][i] = hnew;
                    lds.h[curp ^ 1][row][j] = f2bf(hnew);
                    if constexpr (TRAIN) {
                        // cache stores straight from registers (8-byte,
                        // 128 B per 16-lane group): no LDS staging tile,
                        // no post-barrier round-trip
                        if (!(dbg & 2u)) {
                            bf16x4 pk = {f2bf(r8[mt][i]), f2bf(z8[mt][i]),
                                         f2bf(n8[mt][i]), f2bf(hg8[mt][i])};
                            *reinterpret_cast<bf16x4*>(
                                cache +
                                (((size_t)t * B + b0 + row) * 2 + dir) * 4 * H +
                                4 * j) = pk;
                        }
                    }
                }
        }
        __syncthreads();  // h[curp^1] (+ cache staging) published

        // cooperative wide store of h to hseq (coalesced 16B per lane),
        // overlaps the next step's MFMAs as plain VMEM traffic
        if (!(dbg & 1u)) {
            bf16* dst = hseq + (((size_t)t * B + b0) * 2 + dir) * H;
            const int e = tid * CW;
            const int row = e / H, col = e % H;
            *reinterpret_cast<xvec*>(dst + (size_t)row * 2 * H + col) =
                *reinterpret_cast<const xvec*>(&lds.h[curp ^ 1][row][col]);
        }
    };

    int ti = 0;
    for (; ti + 2 <= T; ti += 2) {
        body(ti, 0, stageA, stageB);
        body(ti + 1, 1, stageB, stageA);
    }
    if (ti < T) body(ti, 0, stageA, stageB);
}

void gru_layer_fwd(const void* xg, const void* u, const float* bhh, void* hseq,
                   void* cache, int T, int B, hipStream_t stream,
                   uint32_t dbg) {
    dim3 grid(B / MB, 2);
    dim3 block(WAVES * 64);
    // MBT=16 halves each thread's gate-chain count — the gate phase sits
    // at the 2-wave VALU-issue floor, so the wall nearly halves (eval
    // 175.6 -> 97.7 us measured) and the WG count doubles (serving
    // packing). ROKO_GRU_MB=32 reverts both paths.
    static const int mb_sel = [] {
        const char* f = getenv("ROKO_GRU_MB");
        return (f && std::string(f) == "32") ? 32 : 16;
    }();
    if (cache) {
        if (mb_sel == 16 && (B % 16) == 0)
            hipLaunchKernelGGL((gru_layer_fwd_kernel<true, 0, 16>),
                               dim3(B / 16, 2), block, 0, stream,
                               static_cast<const bf16*>(xg),
                               static_cast<const bf16*>(u), bhh,
                               static_cast<bf16*>(hseq),
                               static_cast<bf16*>(cache), T, B, dbg, nullptr,
                               nullptr, nullptr, 0);
        else
            hipLaunchKernelGGL((gru_layer_fwd_kernel<true, 0>), grid, block,
                               0, stream, static_cast<const bf16*>(xg),
                               static_cast<const bf16*>(u), bhh,
                               static_cast<bf16*>(hseq),
                               static_cast<bf16*>(cache), T, B, dbg, nullptr,
                               nullptr, nullptr, 0);
    }
    else {
        // eval tile choice: MBT=16 doubles the WG count and halves each
        // thread's gate-chain count (ROKO_GRU_MB=32 reverts)
        static const int mb_eval = [] {
            const char* f = getenv("ROKO_GRU_MB");
            return (f && std::string(f) == "32") ? 32 : 16;
        }();
        if (mb_eval == 16 && (B % 16) == 0)
            hipLaunchKernelGGL((gru_layer_fwd_kernel<false, 0, 16>),
                               dim3(B / 16, 2), block, 0, stream,
                               static_cast<const bf16*>(xg),
                               static_cast<const bf16*>(u), bhh,
                               static_cast<bf16*>(hseq), nullptr, T, B, dbg,
                               nullptr, nullptr, nullptr, 0);
        else
            hipLaunchKernelGGL((gru_layer_fwd_kernel<false, 0>), grid, block,
                               0, stream, static_cast<const bf16*>(xg),
                               static_cast<const bf16*>(u), bhh,
                               static_cast<bf16*>(hseq), nullptr, T, B, dbg,
                               nullptr, nullptr, nullptr, 0);
    }
}

// serving variant: xg computed in-kernel from (x, W_ih, b_ih); `xg` is a
// (T, B, 2, 3H) workspace
void gru_layer_fwd_fused(const void* x, const void* w_ih_p, const void* b_ih,
                         const void* u, const float* bhh, void* xg_ws,
                         void* hseq, int T, int B, int IN, int KP,
                         hipStream_t stream) {
    dim3 grid(B / MB, 2);
    dim3 block(WAVES * 64);
    if (KP == 512)
        hipLaunchKernelGGL((gru_layer_fwd_kernel<false, 512>), grid, block, 0,
                           stream, static_cast<bf16*>(xg_ws),
                           static_cast<const bf16*>(u),
                           static_cast<const float*>(bhh),
                           static_cast<bf16*>(hseq), nullptr, T, B, 0u,
                           static_cast<const bf16*>(x),
                           static_cast<const bf16*>(w_ih_p),
                           static_cast<const bf16*>(b_ih), IN);
    else if (KP == 256)
        hipLaunchKernelGGL((gru_layer_fwd_kernel<false, 256>), grid, block, 0,
                           stream, static_cast<bf16*>(xg_ws),
                           static_cast<const bf16*>(u),
                           static_cast<const float*>(bhh),
                           static_cast<bf16*>(hseq), nullptr, T, B, 0u,
                           static_cast<const bf16*>(x),
                           static_cast<const bf16*>(w_ih_p),
                           static_cast<const bf16*>(b_ih), IN);
    else
        (void)0;  // unsupported KP: caller validates (bindings.cpp)
}

// ---------------------------------------------------------------------------
// Backward (BPTT) — the sequential part only. The kernel walks the T steps
// in reverse, carrying dL/dh in registers and emitting per-step gate
// gradients dg = [dxr, dxz, dxn, dhgn]; the batched weight/input gradient
// reductions (dU = dhg^T·h_prev, dW_ih = dxg^T·x, dx = dxg·W_ih) are plain
// GEMMs done host-side with hipBLASLt.
//
// Math per step (forward: r,z = sig, n = tanh(xn + r*hgn), h' = (1-z)n + zh):
//   dn   = dh * (1-z)            dz  = dh * (h_prev - n)
//   dan  = dn * (1 - n^2)        dxn = dan         dhgn = dan * r
//   dr   = dan * hgn             dxr = dhgr = dr * r * (1-r)
//   dxz  = dhgz = dz * z * (1-z)
//   dh_prev = dh * z + dhg · U
// ---------------------------------------------------------------------------

constexpr int BW_WAVES = 8;  // dhg GEMM has K=384: 24 MFMA/wave at 8 waves
constexpr int BWCH_C = 4;    // cache chunks per thread (32x512 / 512 / 8)

template <int MBT = MB>
__global__ __launch_bounds__(BW_WAVES * 64, 2) void gru_layer_bwd_kernel(
    const bf16* __restrict__ cache,  // (T, B, 2, H, 4) [r z n hgn] packed
    const bf16* __restrict__ hseq,   // (T, B, 2, H)
    const bf16* __restrict__ dhin,   // (T, B, 2, H) grad wrt layer output
    const bf16* __restrict__ ut,     // (2, H, 3H) = weight_hh^T
    bf16* __restrict__ dxg,          // (T, B, 2, 3H) out: [dxr dxz dxn] —
                                     // (T*B, 768) view feeds dW_ih/dx GEMMs
    bf16* __restrict__ dhg,          // (2, T, B, 3H) out: [dxr dxz dhgn]
                                     // dir-major so each dir slice is a
                                     // contiguous (T*B, 384) GEMM operand
    int T, int B,
    uint32_t dbg) {  // timing bisection: 1 no global stores, 2 no gate
                     // VALU, 4 no MFMA, 8 no staging
    // ONE barrier per step (was two): dhg is double-buffered so the next
    // step's gate writes never touch the tile the current step's dh GEMM
    // and output stores still read, and the dg_st staging tile is gone —
    // dxg/dhg stores read straight from lds.dhg (dxr/dxz/dhgn blocks) plus
    // per-lane register stores for the dxn column block.
    constexpr int MT = MBT / 16;               // 16-row m-subtiles
    constexpr int BWCH = MBT * 4 * H / (BW_WAVES * 64 * 8);  // cache chunks
    __shared__ struct {
        bf16 cache_st[2][MBT][4 * H];  // double-buffered staged cache[t]
        bf16 dhin_st[2][MBT][H];       // double-buffered staged dhin[t]
        bf16 hprev_st[2][MBT][H];      // double-buffered staged h_{t-1}
        bf16 dhg[2][MBT][G3 + 8];      // A-operand of the dh GEMM (2 bufs)
    } lds;

    const int dir = blockIdx.y;
    const int b0 = blockIdx.x * MBT;
    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int j0 = wid * 16;
    const int lrow = lane >> 4;
    const int lcol = lane & 15;
    if (wid >= BW_WAVES / 2) __builtin_amdgcn_s_setprio(1);

    // B-fragments of U for dh_prev = dhg·U: B[k=j][col=khid] = Ut[khid][j]
    bf16x8 ufrag[12];
#pragma unroll
    for (int kb = 0; kb < 12; ++kb) {
        const int col = j0 + lcol;          // this wave's hidden column
        const int j = kb * 32 + 8 * lrow;   // gate-row index
        ufrag[kb] = *reinterpret_cast<const bf16x8*>(
            ut + ((size_t)dir * H + col) * G3 + j);
    }

    float dhc[MT][4];  // dh carry, fragment-shaped
#pragma unroll
    for (int mt = 0; mt < MT; ++mt)
#pragma unroll
        for (int i = 0; i < 4; ++i) dhc[mt][i] = 0.0f;

    const int stp = (dir == 0) ? -1 : 1;     // BPTT walks t backwards
    const int t0 = (dir == 0) ? T - 1 : 0;
    auto t_of = [&](int sidx) { return t0 + stp * sidx; };

    // register-staged loads for one step: cache (4 chunks) + dhin + hprev
    auto issue_loads = [&](int sidx, bf16x8 (&rc)[BWCH], bf16x8& rdh,
                           bf16x8& rhp) {
        if (dbg & 8u) return;
        const int t = t_of(sidx);
        const int tp = (dir == 0) ? t - 1 : t + 1;
        const bf16* src = cache + (((size_t)t * B + b0) * 2 + dir) * 4 * H;
        const int row4 = tid / 64, col4 = (tid % 64) * 8;
#pragma unroll
        for (int q = 0; q < BWCH; ++q)
            rc[q] = *reinterpret_cast<const bf16x8*>(
                src + (size_t)(row4 + q * 8) * 2 * 4 * H + col4);
        const int row = tid / 16, col = (tid % 16) * 8;
        rhp = bf16x8{};
        if (row < MBT) {
            rdh = *reinterpret_cast<const bf16x8*>(
                dhin + (((size_t)t * B + b0) * 2 + dir) * H + (size_t)row * 2 * H + col);
            if (tp >= 0 && tp < T)
                rhp = *reinterpret_cast<const bf16x8*>(
                    hseq + (((size_t)tp * B + b0) * 2 + dir) * H + (size_t)row * 2 * H + col);
        } else
            rdh = bf16x8{};
    };
    auto write_stage = [&](int buf, bf16x8 (&rc)[BWCH], bf16x8& rdh,
                           bf16x8& rhp) {
        const int row4 = tid / 64, col4 = (tid % 64) * 8;
#pragma unroll
        for (int q = 0; q < BWCH; ++q)
            *reinterpret_cast<bf16x8*>(&lds.cache_st[buf][row4 + q * 8][col4]) = rc[q];
        const int row = tid / 16, col = (tid % 16) * 8;
        if (row < MBT) {
            *reinterpret_cast<bf16x8*>(&lds.dhin_st[buf][row][col]) = rdh;
            *reinterpret_cast<bf16x8*>(&lds.hprev_st[buf][row][col]) = rhp;
        }
    };

    bf16x8 rcA[BWCH], rcB[BWCH];
    bf16x8 rdhA, rhpA, rdhB, rhpB;
    // prologue: step 0 straight to LDS buffer 0; step 1 to registers A
    issue_loads(0, rcA, rdhA, rhpA);
    write_stage(0, rcA, rdhA, rhpA);
    if (T > 1) issue_loads(1, rcA, rdhA, rhpA);
    __syncthreads();

    auto body = [&](int sidx, int curp, bf16x8 (&rc_wr)[BWCH], bf16x8& rdh_wr,
                    bf16x8& rhp_wr, bf16x8 (&rc_ld)[BWCH], bf16x8& rdh_ld,
                    bf16x8& rhp_ld) {
        const int t = t_of(sidx);
        // issue step s+2 loads FIRST, then write step s+1's staged registers
        if (sidx + 2 < T) issue_loads(sidx + 2, rc_ld, rdh_ld, rhp_ld);
        if (sidx + 1 < T) write_stage(curp ^ 1, rc_wr, rdh_wr, rhp_wr);

        // ---- gate gradients from stage[curp] ------------------------------
        // reads batched first, then 8 independent arithmetic chains, then
        // the stores — same restructure as the forward's gate phase (the
        // interleaved per-element form serialized the read latencies)
        float dhp_part[MT][4];
        float dxn8[MT][4];  // kept in registers for the direct dxg store
        {
            const int j = j0 + lcol;
            float dh8[MT][4], hp8[MT][4];
            bf16x4 pk8[MT][4];
#pragma unroll
            for (int mt = 0; mt < MT; ++mt)
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    const int row = mt * 16 + lrow * 4 + i;
                    dh8[mt][i] =
                        dhc[mt][i] + bf2f(lds.dhin_st[curp][row][j]);
                    pk8[mt][i] = *reinterpret_cast<const bf16x4*>(
                        &lds.cache_st[curp][row][4 * j]);
                    hp8[mt][i] = bf2f(lds.hprev_st[curp][row][j]);
                }
            if (dbg & 2u) {  // timing: gate math stripped
#pragma unroll
                for (int mt = 0; mt < MT; ++mt)
#pragma unroll
                    for (int i = 0; i < 4; ++i) {
                        const int row = mt * 16 + lrow * 4 + i;
                        const float v = dh8[mt][i] + hp8[mt][i];
                        dhp_part[mt][i] = v;
                        dxn8[mt][i] = v;
                        lds.dhg[curp][row][0 * H + j] = f2bf(v);
                        lds.dhg[curp][row][1 * H + j] = f2bf(v);
                        lds.dhg[curp][row][2 * H + j] = f2bf(v);
                    }
            } else
#pragma unroll
            for (int mt = 0; mt < MT; ++mt)
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    const int row = mt * 16 + lrow * 4 + i;
                    const float dh = dh8[mt][i];
                    const float r = bf2f(pk8[mt][i][0]);
                    const float z = bf2f(pk8[mt][i][1]);
                    const float n = bf2f(pk8[mt][i][2]);
                    const float hgn = bf2f(pk8[mt][i][3]);
                    const float dn = dh * (1.0f - z);
                    const float dz = dh * (hp8[mt][i] - n);
                    const float dan = dn * (1.0f - n * n);
                    const float dhgn = dan * r;
                    const float dxr = dan * hgn * r * (1.0f - r);
                    const float dxz = dz * z * (1.0f - z);
                    dhp_part[mt][i] = dh * z;
                    dxn8[mt][i] = dan;
                    lds.dhg[curp][row][0 * H + j] = f2bf(dxr);
                    lds.dhg[curp][row][1 * H + j] = f2bf(dxz);
                    lds.dhg[curp][row][2 * H + j] = f2bf(dhgn);
                }
        }
        __syncthreads();  // dhg[curp] and stage[curp^1] published (the ONLY
                          // barrier per step; next step writes dhg[curp^1])

        // ---- dh_prev = dh*z + dhg · U  (24 MFMA per wave) -----------------
        f32x4 acc[MT];
#pragma unroll
        for (int mt = 0; mt < MT; ++mt) acc[mt] = f32x4{0.f, 0.f, 0.f, 0.f};
        if (!(dbg & 4u)) {
#pragma unroll
            for (int kb = 0; kb < 12; ++kb) {
#pragma unroll
                for (int mt = 0; mt < MT; ++mt) {
                    bf16x8 a = lds_load_a_frag(&lds.dhg[curp][0][0], mt * 16,
                                               kb * 32, G3 + 8);
                    acc[mt] = mfma16x16x32(a, ufrag[kb], acc[mt]);
                }
            }
        }
#pragma unroll
        for (int mt = 0; mt < MT; ++mt)
#pragma unroll
            for (int i = 0; i < 4; ++i) dhc[mt][i] = dhp_part[mt][i] + acc[mt][i];

        // ---- store dxg + dhg tiles (GEMM-ready layouts, no host cats) -----
        // dxg = [dxr dxz dxn]: the dxr/dxz blocks copy wide from lds.dhg,
        // dxn comes from this lane's registers (scalar stores)
        if (!(dbg & 1u)) {
            bf16* dst = dxg + (((size_t)t * B + b0) * 2 + dir) * G3;
            const int row = tid / 32, col = (tid % 32) * 8;  // 2H block
#pragma unroll
            for (int q = 0; q < MT; ++q)
                *reinterpret_cast<bf16x8*>(
                    dst + (size_t)(row + q * 16) * 2 * G3 + col) =
                    *reinterpret_cast<const bf16x8*>(
                        &lds.dhg[curp][row + q * 16][col]);
            const int j = j0 + lcol;
#pragma unroll
            for (int mt = 0; mt < MT; ++mt)
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    const int row2 = mt * 16 + lrow * 4 + i;
                    dst[(size_t)row2 * 2 * G3 + 2 * H + j] =
                        f2bf(dxn8[mt][i]);
                }
        }
        if (!(dbg & 1u)) {
            // dhg tensor = [dxr dxz dhgn] — exactly lds.dhg's live columns
            bf16* dst = dhg + (((size_t)dir * T + t) * B + b0) * G3;
            const int row = tid / 64, col = (tid % 64) * 8;
            if (col < G3) {
#pragma unroll
                for (int q = 0; q < MBT / 8; ++q)
                    *reinterpret_cast<bf16x8*>(
                        dst + (size_t)(row + q * 8) * G3 + col) =
                        *reinterpret_cast<const bf16x8*>(
                            &lds.dhg[curp][row + q * 8][col]);
            }
        }
    };

    int sidx = 0;
    for (; sidx + 2 <= T; sidx += 2) {
        body(sidx, 0, rcA, rdhA, rhpA, rcB, rdhB, rhpB);
        body(sidx + 1, 1, rcB, rdhB, rhpB, rcA, rdhA, rhpA);
    }
    if (sidx < T) body(sidx, 0, rcA, rdhA, rhpA, rcB, rdhB, rhpB);
}

void gru_layer_bwd(const void* cache, const void* hseq, const void* dhin,
                   const void* ut, void* dxg, void* dhg, int T, int B,
                   hipStream_t stream, uint32_t dbg) {
    dim3 block(BW_WAVES * 64);
    static const int mb_sel = [] {
        const char* f = getenv("ROKO_GRU_MB");
        return (f && std::string(f) == "32") ? 32 : 16;
    }();
    if (mb_sel == 16 && (B % 16) == 0)
        hipLaunchKernelGGL((gru_layer_bwd_kernel<16>), dim3(B / 16, 2), block,
                           0, stream, static_cast<const bf16*>(cache),
                           static_cast<const bf16*>(hseq),
                           static_cast<const bf16*>(dhin),
                           static_cast<const bf16*>(ut),
                           static_cast<bf16*>(dxg), static_cast<bf16*>(dhg),
                           T, B, dbg);
    else
        hipLaunchKernelGGL((gru_layer_bwd_kernel<32>), dim3(B / MB, 2), block,
                           0, stream, static_cast<const bf16*>(cache),
                           static_cast<const bf16*>(hseq),
                           static_cast<const bf16*>(dhin),
                           static_cast<const bf16*>(ut),
                           static_cast<bf16*>(dxg), static_cast<bf16*>(dhg),
                           T, B, dbg);
}

}  // namespace rk
