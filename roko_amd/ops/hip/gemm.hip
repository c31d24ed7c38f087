// bf16 GEMM + bias for the GRU input-projection shape family
// (M ~ 11520 = T*B, N <= 768, K <= 768) on gfx950.
//
// hipBLASLt's heuristic picks ~110 TF/s tiles for these shapes (xg GEMM
// measured 81 us — profiles/train_r01_kernel_stats.txt); the MFMA floor is
// ~10x lower. Canonical CDNA GEMM structure (cdna_hip_programming.md §5):
// 256x64 C-tiles, K staged in 32-wide slices through double-buffered LDS,
// A-fragments ds_read_b128 from row-major [row][k] images, B transposed at
// staging time into [col][k] images, bias fused into the epilogue, output
// staged through LDS for 16-byte coalesced stores.
//
//   C (M, N) bf16 = A (M, K) bf16 · B (K, N) bf16 + bias (N) f32
//
// M/N/K need not be multiples of the tile sizes (tails are masked).

#include <cstdint>

#include "common.h"

namespace rk {
namespace gemm {

constexpr int BM = 256;  // C-tile rows per workgroup
constexpr int BN = 64;   // C-tile cols per workgroup
constexpr int BK = 32;   // K slice per stage
constexpr int LDA = BK + 8;   // LDS row strides (bank-conflict pad)
constexpr int WAVES = 8;

__global__ __launch_bounds__(WAVES * 64, 2) void gemm_bias_kernel(
    const bf16* __restrict__ A,   // (M, K) row-major
    const bf16* __restrict__ B,   // (K, N) row-major
    const float* __restrict__ bias,  // (N) or nullptr
    bf16* __restrict__ C,         // (M, N) row-major
    int M, int N, int K) {
    __shared__ struct {
        bf16 a[2][BM][LDA];   // [row][k]
        bf16 bt[2][BN][LDA];  // [col][k] (transposed at staging)
    } lds;

    const int m0 = blockIdx.x * BM;
    const int n0 = blockIdx.y * BN;
    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int lrow = lane >> 4;
    const int lcol = lane & 15;

    // wave tiling: 8 waves x (2 m-subtiles x 4 n-subtiles)? -> each wave owns
    // two 16-row strips across two 16-col strips: tiles (mt, nt) with
    // mt = wid * 2 + {0,1} over 16 m-subtiles, nt = {0..3}
    // accumulators: 2 mt x 4 nt = 8 fragments
    f32x4 acc[2][4];
#pragma unroll
    for (int a_ = 0; a_ < 2; ++a_)
#pragma unroll
        for (int b_ = 0; b_ < 4; ++b_) acc[a_][b_] = f32x4{0.f, 0.f, 0.f, 0.f};

    const int kq = (K + BK - 1) / BK;

    // cooperative staging: A tile BM x BK (8192 elems = 16/thread = 2 x b128),
    // B tile BK x BN (2048 elems = 4/thread, transposed scalar writes)
    // b128 loads/stores need 16-byte alignment: row starts are only aligned
    // when the row stride is a multiple of 8 bf16 elements (live shapes like
    // K=500 are not) — take the scalar tail path otherwise
    const bool a_aligned = (K % 8) == 0;
    auto stage = [&](int buf, int k0) {
        const bool full_k = (k0 + BK <= K);
#pragma unroll
        for (int p = 0; p < 2; ++p) {
            const int e = (p * 512 + tid) * 8;
            const int r = e / BK, kk = e % BK;
            const int row = m0 + r;
            bf16 v[8];
            if (full_k && a_aligned && row < M) {
                *reinterpret_cast<bf16x8*>(v) =
                    *reinterpret_cast<const bf16x8*>(A + (size_t)row * K + k0 + kk);
            } else {
#pragma unroll
                for (int q = 0; q < 8; ++q)
                    v[q] = (row < M && k0 + kk + q < K)
                               ? A[(size_t)row * K + k0 + kk + q] : f2bf(0.f);
            }
            *reinterpret_cast<bf16x8*>(&lds.a[buf][r][kk]) =
                *reinterpret_cast<const bf16x8*>(v);
        }
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            const int e = p * 512 + tid;
            const int kk = e / BN, c = e % BN;
            bf16 v = f2bf(0.f);
            if (k0 + kk < K && n0 + c < N) v = B[(size_t)(k0 + kk) * N + n0 + c];
            lds.bt[buf][c][kk] = v;
        }
    };

    stage(0, 0);
    __syncthreads();

    for (int q = 0; q < kq; ++q) {
        if (q + 1 < kq) stage((q + 1) & 1, (q + 1) * BK);
        bf16x8 bfr[4];
#pragma unroll
        for (int b_ = 0; b_ < 4; ++b_)
            bfr[b_] = lds_load_b_frag_t(&lds.bt[q & 1][0][0], b_ * 16, 0, LDA);
#pragma unroll
        for (int a_ = 0; a_ < 2; ++a_) {
            const int mt = wid * 2 + a_;
            bf16x8 af = lds_load_a_frag(&lds.a[q & 1][0][0], mt * 16, 0, LDA);
#pragma unroll
            for (int b_ = 0; b_ < 4; ++b_)
                acc[a_][b_] = mfma16x16x32(af, bfr[b_], acc[a_][b_]);
        }
        __syncthreads();
    }

    // epilogue: bias + staged coalesced store (reuse the A buffers)
    bf16* cst = &lds.a[0][0][0];  // BM x BN staging, stride BN
#pragma unroll
    for (int a_ = 0; a_ < 2; ++a_) {
        const int mt = wid * 2 + a_;
#pragma unroll
        for (int b_ = 0; b_ < 4; ++b_) {
            const int c = b_ * 16 + lcol;
            const float bv = bias ? bias[min(n0 + c, N - 1)] : 0.f;
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int r = mt * 16 + lrow * 4 + i;
                cst[r * BN + c] = f2bf(acc[a_][b_][i] + bv);
            }
        }
    }
    __syncthreads();
    {
        const int n_full = (n0 + BN <= N) && ((N % 8) == 0);
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            const int e = (p * 512 + tid) * 8;
            const int r = e / BN, c = e % BN;
            const int row = m0 + r;
            if (row < M) {
                if (n_full) {
                    *reinterpret_cast<bf16x8*>(C + (size_t)row * N + n0 + c) =
                        *reinterpret_cast<const bf16x8*>(&cst[r * BN + c]);
                } else {
#pragma unroll
                    for (int u = 0; u < 8; ++u)
                        if (n0 + c + u < N)
                            C[(size_t)row * N + n0 + c + u] = cst[r * BN + c + u];
                }
            }
        }
    }
}

}  // namespace gemm

void gemm_bias(const void* A, const void* B, const float* bias, void* C,
               int M, int N, int K, hipStream_t stream) {
    dim3 grid((M + gemm::BM - 1) / gemm::BM, (N + gemm::BN - 1) / gemm::BN);
    hipLaunchKernelGGL(gemm::gemm_bias_kernel, grid, dim3(gemm::WAVES * 64), 0,
                       stream, static_cast<const bf16*>(A),
                       static_cast<const bf16*>(B), bias, static_cast<bf16*>(C),
                       M, N, K);
}

// ---------------------------------------------------------------------------
// split-K  C (M, N) f32  +=  A (K, M)^T · B (K, N),  bf16 inputs
// ---------------------------------------------------------------------------
// The GRU weight-gradient reductions (dU = dhg^T·h_prev at (384,128) and
// dW_ih = dxg^T·x at (768,500), both K = T*B ~ 11520) map to transpose-A
// GEMMs that hipBLASLt runs WITHOUT split-K — 6 workgroups on a 256-CU chip,
// 81 us each (profiles/train_r01_latest). Here K is sliced across the grid
// (one 64-row stage per iteration, both operands transposed into LDS at
// staging time) and every workgroup commits its partial tile with relaxed
// agent-scope f32 atomics.

namespace gemmatb {

constexpr int BM = 128;  // C rows (= A cols) per workgroup
constexpr int BN = 128;  // C cols per workgroup
constexpr int BK = 64;   // K slice per stage
constexpr int KSLICE = 256;  // K rows per workgroup (KSLICE/BK stages)
constexpr int LDM = BM + 8;  // [k][m] row stride (staged UNtransposed)
constexpr int WAVES = 8;

__global__ __launch_bounds__(WAVES * 64, 2) void atb_splitk_kernel(
    const bf16* __restrict__ A,  // (K, M) rows strided by lda
    const bf16* __restrict__ B,  // (K, N) rows strided by ldb
    float* __restrict__ ws,      // (n_slices, M, N) partials (plain stores —
                                 // the atomic-commit version spent ~50 us in
                                 // 45-way same-address contention on dW_ih)
    int M, int N, int K, int lda, int ldb) {
    // staged in the GLOBAL orientation [k][m] / [k][n]: the transposed
    // staging this replaced wrote 8 scalar u16 per thread at an 8-row lane
    // stride whose dword period gcd'd with the 64 banks to 32 — 8-way bank
    // conflicts on every write, 88% LDS-conflict rate and 7.6% ACTIVE in
    // PMC (profiles/pmc_train_r01.txt). Un-transposed staging is straight
    // b128 stores; the fragment reads below pay scalar u16 loads instead
    // (2-cyc issues that hide under the MFMAs — docs/KERNELS.md lesson 3).
    __shared__ struct {
        bf16 at[2][BK][LDM];  // [k][m]
        bf16 bt[2][BK][LDM];  // [k][n]
    } lds;

    const int k_begin = blockIdx.x * KSLICE;
    const int m0 = blockIdx.y * BM;
    const int n0 = blockIdx.z * BN;
    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int lrow = lane >> 4;
    const int lcol = lane & 15;

    // 8x8 = 64 sub-tiles over 8 waves: wave owns one 16-row strip (mt = wid)
    // across all 8 col strips
    f32x4 acc[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) acc[i] = f32x4{0.f, 0.f, 0.f, 0.f};

    // stage one BK-slice: A rows [k0, k0+BK) cols [m0, m0+BM) transposed to
    // [m][k]; same for B. 128*64 elems each = 16/thread.
    // same alignment rule as gemm_bias_kernel: vector loads only when the
    // row stride keeps every row 16-byte aligned
    const bool m_aligned = (lda % 8) == 0;
    const bool n_aligned = (ldb % 8) == 0;
    auto stage = [&](int buf, int k0) {
#pragma unroll
        for (int p = 0; p < 2; ++p) {
            const int e = (p * 512 + tid) * 8;   // element in (BK, BM) slab
            const int kk = e / BM, m = e % BM;   // 8 consecutive m per thread
            bf16 v[8];
            const int krow = k0 + kk;
            if (krow < K && m_aligned && m0 + m + 7 < M) {
                *reinterpret_cast<bf16x8*>(v) = *reinterpret_cast<const bf16x8*>(
                    A + (size_t)krow * lda + m0 + m);
            } else {
#pragma unroll
                for (int q = 0; q < 8; ++q)
                    v[q] = (krow < K && m0 + m + q < M)
                               ? A[(size_t)krow * lda + m0 + m + q] : f2bf(0.f);
            }
            *reinterpret_cast<bf16x8*>(&lds.at[buf][kk][m]) =
                *reinterpret_cast<const bf16x8*>(v);
        }
#pragma unroll
        for (int p = 0; p < 2; ++p) {
            const int e = (p * 512 + tid) * 8;
            const int kk = e / BN, n = e % BN;
            bf16 v[8];
            const int krow = k0 + kk;
            if (krow < K && n_aligned && n0 + n + 7 < N) {
                *reinterpret_cast<bf16x8*>(v) = *reinterpret_cast<const bf16x8*>(
                    B + (size_t)krow * ldb + n0 + n);
            } else {
#pragma unroll
                for (int q = 0; q < 8; ++q)
                    v[q] = (krow < K && n0 + n + q < N)
                               ? B[(size_t)krow * ldb + n0 + n + q] : f2bf(0.f);
            }
            *reinterpret_cast<bf16x8*>(&lds.bt[buf][kk][n]) =
                *reinterpret_cast<const bf16x8*>(v);
        }
    };

    stage(0, k_begin);
    __syncthreads();
    const int stages = KSLICE / BK;
    for (int s = 0; s < stages; ++s) {
        if (s + 1 < stages) stage((s + 1) & 1, k_begin + (s + 1) * BK);
#pragma unroll
        for (int kb = 0; kb < BK / 32; ++kb) {
            bf16x8 af = lds_load_a_frag_t(&lds.at[s & 1][0][0], wid * 16,
                                          kb * 32, LDM);
#pragma unroll
            for (int b_ = 0; b_ < 8; ++b_) {
                bf16x8 bf_ = lds_load_b_frag_km(&lds.bt[s & 1][0][0], kb * 32,
                                                b_ * 16, LDM);
                acc[b_] = mfma16x16x32(af, bf_, acc[b_]);
            }
        }
        __syncthreads();
    }

    // plain-store the partial tile into this slice's workspace slab
    float* slab = ws + (size_t)blockIdx.x * M * N;
#pragma unroll
    for (int b_ = 0; b_ < 8; ++b_) {
        const int n = n0 + b_ * 16 + lcol;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int m = m0 + wid * 16 + lrow * 4 + i;
            if (m < M && n < N) slab[(size_t)m * N + n] = acc[b_][i];
        }
    }
}

// ws (S, MN) -> out (MN): one coalesced sweep
__global__ __launch_bounds__(256) void slice_sum_kernel(
    const float* __restrict__ ws, float* __restrict__ out, int S,
    int64_t MN) {
    const int64_t i0 = ((int64_t)blockIdx.x * 256 + threadIdx.x) * 4;
    if (i0 + 3 >= MN) {
        for (int64_t i = i0; i < MN; ++i) {
            float a = 0.f;
            for (int s = 0; s < S; ++s) a += ws[(size_t)s * MN + i];
            out[i] = a;
        }
        return;
    }
    float a[4] = {0.f, 0.f, 0.f, 0.f};
    for (int s = 0; s < S; ++s) {
        const float* p = ws + (size_t)s * MN + i0;
#pragma unroll
        for (int q = 0; q < 4; ++q) a[q] += p[q];
    }
#pragma unroll
    for (int q = 0; q < 4; ++q) out[i0 + q] = a[q];
}

}  // namespace gemmatb

// ---------------------------------------------------------------------------
// Serving xg projection GEMM:  C (M, 768) = A (M, KP) · B (KP, 768) + bias
// ---------------------------------------------------------------------------
// hipBLASLt runs this shape (M = T*B = 11520, N = 768, K <= 512) at
// ~543 TF/s on ~225 workgroups — ~105 CU·us of aggregate chip time per
// served window, the single largest serving cost after the front
// (profiles/infer_r02_v3_kernel_stats.csv). This kernel trades LDS staging
// for an LDS-FREE, BARRIER-FREE design shaped for the pipelined server:
//   * grid = (M/384) x 6: 180 workgroups — under one full wave of the
//     256-CU chip, so concurrent serving chains fill the leftover CUs;
//   * each wave owns a PRIVATE 48x128 C tile (3 m-subtiles x 8 n-subtiles,
//     24 accumulators in AGPR space) and reads its A/B fragments straight
//     from global (A: aligned b128 rows — the caller pads K to 512/256;
//     B: L2-resident (KP,768) slice reused by the 30 m-workgroups of its
//     n-range), software-pipelined one k-step ahead;
//   * no __syncthreads anywhere: nothing is shared between waves.
namespace xggemm {

// MEASURED NEGATIVE (kept for the record, not wired into serving): both
// tilings lose to hipBLASLt — 48x128/wave: 47.9 us vs addmm 25.6; 32x64:
// 68.9 us. Without LDS staging every wave re-reads its whole B slice from
// L2 (no cross-wave operand reuse), and the compiler's vmcnt scoreboard
// still inserts ~44 full drains per kernel under 180+ live fragment
// registers. The library's LDS-staged tiles win this shape; serving keeps
// torch.addmm (see docs/KERNELS.md lesson 13-15).
constexpr int XN = 768;
constexpr int WM = 32;    // rows per wave (2 m-subtiles)
constexpr int BN = 64;    // cols per workgroup (4 n-subtiles per wave)
constexpr int WAVES = 8;  // 8 waves x 32 rows = 256 rows per workgroup

template <int KP>
__global__ __launch_bounds__(WAVES * 64, 2) void xg_gemm_kernel(
    const bf16* __restrict__ A,     // (M, KP) row-major, 16-B-aligned rows
    const bf16* __restrict__ Bt,    // (768, KP) row-major = B^T (the
                                    // serving cache's padded W_ih image)
    const bf16* __restrict__ bias,  // (768)
    bf16* __restrict__ C,           // (M, 768)
    int M) {
    constexpr int KB = KP / 32;
    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int lrow = lane >> 4;
    const int lcol = lane & 15;
    const int m0 = blockIdx.x * (WAVES * WM) + wid * WM;
    const int n0 = blockIdx.y * BN;

    f32x4 acc[2][4];
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
        for (int nt = 0; nt < 4; ++nt) acc[mt][nt] = f32x4{0.f, 0.f, 0.f, 0.f};

    // fragment loaders (all global; A rows aligned because KP % 8 == 0)
    auto a_frag = [&](int mt, int kb) {
        return global_load_a_frag(A + (size_t)m0 * KP, mt * 16, kb * 32, KP);
    };
    auto b_frag = [&](int nt, int kb) {
        // B^T (768, KP) row-major is the [col][k] layout: the B-fragment
        // load is the A-fragment address pattern — aligned b128 per lane
        return global_load_a_frag(Bt, n0 + nt * 16, kb * 32, KP);
    };

    // two explicit register sets, alternated by an unrolled two-body loop:
    // selecting the sets through references compiled to 44 full vmcnt(0)
    // drains (the compiler loses per-load scoreboard tracking — the same
    // failure gru.hip's stage rotation hit, docs/KERNELS.md lesson 2)
    bf16x8 afA[2], afB[2], bfA[4], bfB[4];
    auto load_set = [&](int kb, bf16x8(&af)[2], bf16x8(&bf)[4]) {
#pragma unroll
        for (int mt = 0; mt < 2; ++mt) af[mt] = a_frag(mt, kb);
#pragma unroll
        for (int nt = 0; nt < 4; ++nt) bf[nt] = b_frag(nt, kb);
    };
    auto mfma_set = [&](bf16x8(&af)[2], bf16x8(&bf)[4]) {
#pragma unroll
        for (int mt = 0; mt < 2; ++mt)
#pragma unroll
            for (int nt = 0; nt < 4; ++nt)
                acc[mt][nt] = mfma16x16x32(af[mt], bf[nt], acc[mt][nt]);
    };
    load_set(0, afA, bfA);
#pragma unroll
    for (int kb = 0; kb < KB; kb += 2) {
        if (kb + 1 < KB) load_set(kb + 1, afB, bfB);
        mfma_set(afA, bfA);
        if (kb + 2 < KB) load_set(kb + 2, afA, bfA);
        if (kb + 1 < KB) mfma_set(afB, bfB);
    }

    // epilogue: + bias, scalar bf16 stores (16-lane 32-B runs; the C rows
    // this wave owns are private, no staging needed)
    float br[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) br[nt] = bf2f(bias[n0 + nt * 16 + lcol]);
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
        for (int nt = 0; nt < 4; ++nt)
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int row = m0 + mt * 16 + lrow * 4 + i;
                if (row < M)
                    C[(size_t)row * XN + n0 + nt * 16 + lcol] =
                        f2bf(acc[mt][nt][i] + br[nt]);
            }
}

}  // namespace xggemm

// ---------------------------------------------------------------------------
// xg_gemm2: the LDS-staged, shape-specialized retry. gemm_bias (the generic
// kernel) compiles its runtime-K loop into per-chunk exec-mask branches and
// a vmcnt(0) drain after every staging load (ISA audit in the r2 notes) —
// 119 us for this shape. Here K is a template constant (fully unrolled,
// register-set double-buffering), staging is branchless b64/b128 with no
// bounds masks (M % 256 == 0, N = 768 exactly), and B stages straight from
// the (768, KP) padded W image whose [n][k] rows already ARE the
// b-fragment layout (no transpose, no conflicts).
namespace xg2 {

constexpr int XN = 768;
constexpr int BM = 256;   // C rows per workgroup (2 m-subtiles per wave)
constexpr int BN = 128;   // C cols per workgroup (8 n-subtiles)
constexpr int LD = 40;    // LDS k-slice stride (32 + 8 pad)
constexpr int WAVES = 8;

// phase timing (wave 0, workgroup 0 only): cycles in [commit+issue, mfma,
// barrier, epilogue] accumulated into timing[0..3] when non-null
#define XG2_T0 unsigned long long tp0 = (timing && tid == 0) \
        ? __builtin_amdgcn_s_memtime() : 0
#define XG2_T1(i) if (timing && tid == 0) { \
        unsigned long long tn = __builtin_amdgcn_s_memtime(); \
        tacc[i] += tn - tp0; tp0 = tn; }

template <int KP, int KREAL>
__global__ __launch_bounds__(WAVES * 64, 1) void xg_gemm2_kernel(
    const bf16* __restrict__ A,     // (M, KREAL) row-major (8-B aligned rows)
    const bf16* __restrict__ Bt,    // (768, KP) row-major = B^T, zero-padded
    const bf16* __restrict__ bias,  // (768)
    bf16* __restrict__ C,           // (M, 768)
    int M, unsigned long long* timing = nullptr) {
    unsigned long long tacc[4] = {0, 0, 0, 0};
    // 64-wide k-slices: half the barrier count of the 32-wide version
    // (phase timing showed ~480 cyc/iter of barrier wave-skew), two
    // 16-MFMA bursts per slice.
    constexpr int BK = 64;
    constexpr int KB = KP / BK;
    constexpr int LDK = BK + 8;
    __shared__ struct {
        bf16 a[2][BM][LDK];   // [row][k]
        bf16 b[2][BN][LDK];   // [n][k]
    } lds;

    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int lrow = lane >> 4;
    const int lcol = lane & 15;
    const int m0 = blockIdx.x * BM;
    const int n0 = blockIdx.y * BN;

    f32x4 acc[2][8];
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
        for (int nt = 0; nt < 8; ++nt) acc[mt][nt] = f32x4{0.f, 0.f, 0.f, 0.f};

    // staging, branchless: A 256x64 = 32 el/thread as 8 b64 (8-B row
    // alignment holds for KREAL % 4 == 0), B 128x64 = 16 el/thread as
    // 2 b128 (KP % 8 == 0). Two register sets; the loop runs the MFMA
    // burst FIRST and commits after it, so the commit's vmcnt wait sits
    // a full ~1100-cycle MFMA burst after the loads were issued — the
    // 3-set variant bought the same cover with 24 more registers.
    uint64_t ra[2][8];
    bf16x8 rb[2][2];
    const int a_row = tid >> 1, a_c0 = (tid & 1) * 32;     // 2 threads/row
    const int b_row = tid >> 2, b_c0 = (tid & 3) * 16;     // 4 threads/row
    auto issue = [&](int kq, uint64_t(&va)[8], bf16x8(&vb)[2]) {
        const int k0 = kq * BK;
#pragma unroll
        for (int q = 0; q < 8; ++q) {
            const int col = k0 + a_c0 + q * 4;
            va[q] = (KREAL == KP || col + 4 <= KREAL)
                        ? *reinterpret_cast<const uint64_t*>(
                              A + (size_t)(m0 + a_row) * KREAL + col)
                        : 0u;
        }
#pragma unroll
        for (int j = 0; j < 2; ++j)
            vb[j] = *reinterpret_cast<const bf16x8*>(
                Bt + (size_t)(n0 + b_row) * KP + k0 + b_c0 + j * 8);
    };
    auto commit = [&](int buf, uint64_t(&va)[8], bf16x8(&vb)[2]) {
#pragma unroll
        for (int q = 0; q < 8; ++q)
            *reinterpret_cast<uint64_t*>(&lds.a[buf][a_row][a_c0 + q * 4]) =
                va[q];
#pragma unroll
        for (int j = 0; j < 2; ++j)
            *reinterpret_cast<bf16x8*>(&lds.b[buf][b_row][b_c0 + j * 8]) =
                vb[j];
    };
    auto mfma_step = [&](int buf) {
        // two k-halves; within each, batch the fragment reads before the
        // MFMA burst so the lgkmcnt waits overlap
#pragma unroll
        for (int kh = 0; kh < 2; ++kh) {
            bf16x8 af[2], bfr[8];
#pragma unroll
            for (int mt = 0; mt < 2; ++mt)
                af[mt] = lds_load_a_frag(&lds.a[buf][0][0],
                                         (wid * 2 + mt) * 16, kh * 32, LDK);
#pragma unroll
            for (int nt = 0; nt < 8; ++nt)
                bfr[nt] = lds_load_b_frag_t(&lds.b[buf][0][0], nt * 16,
                                            kh * 32, LDK);
#pragma unroll
            for (int mt = 0; mt < 2; ++mt)
#pragma unroll
                for (int nt = 0; nt < 8; ++nt)
                    acc[mt][nt] = mfma16x16x32(af[mt], bfr[nt], acc[mt][nt]);
        }
    };

    issue(0, ra[0], rb[0]);
    commit(0, ra[0], rb[0]);
    if (KB > 1) issue(1, ra[1], rb[1]);
    __syncthreads();
    XG2_T0;
#pragma unroll
    for (int kq = 0; kq < KB; ++kq) {
        mfma_step(kq & 1);
        XG2_T1(1);
        if (kq + 1 < KB) commit((kq + 1) & 1, ra[(kq + 1) & 1],
                                rb[(kq + 1) & 1]);
        if (kq + 2 < KB) issue(kq + 2, ra[kq & 1], rb[kq & 1]);
        XG2_T1(0);
        __syncthreads();
        XG2_T1(2);
    }

    // epilogue: bias + direct stores from the accumulators. A quarter-wave
    // writes 16 consecutive bf16 (one 32-B chunk) per fragment row; no LDS
    // round-trip (the staged version cost ~4k cycles of exposed latency).
    float br[8];
#pragma unroll
    for (int nt = 0; nt < 8; ++nt)
        br[nt] = bf2f(bias[n0 + nt * 16 + lcol]);
#pragma unroll
    for (int mt = 0; mt < 2; ++mt) {
        const size_t rbase = (size_t)(m0 + (wid * 2 + mt) * 16 + lrow * 4);
#pragma unroll
        for (int nt = 0; nt < 8; ++nt)
#pragma unroll
            for (int i = 0; i < 4; ++i)
                C[(rbase + i) * XN + n0 + nt * 16 + lcol] =
                    f2bf(acc[mt][nt][i] + br[nt]);
    }
    XG2_T1(3);
    if (timing && tid == 0 && blockIdx.x == 0 && blockIdx.y == 0)
        for (int i = 0; i < 4; ++i) timing[i] = tacc[i];
}

}  // namespace xg2

void xg_gemm2(const void* A, const void* Bt, const void* bias, void* C,
              int M, int KREAL, int KP, hipStream_t stream,
              unsigned long long* timing) {
    dim3 grid(M / xg2::BM, xg2::XN / xg2::BN);
    dim3 block(xg2::WAVES * 64);
    if (KP == 512 && KREAL == 500)
        hipLaunchKernelGGL((xg2::xg_gemm2_kernel<512, 500>), grid, block, 0,
                           stream, static_cast<const bf16*>(A),
                           static_cast<const bf16*>(Bt),
                           static_cast<const bf16*>(bias),
                           static_cast<bf16*>(C), M, timing);
    else if (KP == 512 && KREAL == 512)
        hipLaunchKernelGGL((xg2::xg_gemm2_kernel<512, 512>), grid, block, 0,
                           stream, static_cast<const bf16*>(A),
                           static_cast<const bf16*>(Bt),
                           static_cast<const bf16*>(bias),
                           static_cast<bf16*>(C), M, timing);
    else if (KP == 256 && KREAL == 256)
        hipLaunchKernelGGL((xg2::xg_gemm2_kernel<256, 256>), grid, block, 0,
                           stream, static_cast<const bf16*>(A),
                           static_cast<const bf16*>(Bt),
                           static_cast<const bf16*>(bias),
                           static_cast<bf16*>(C), M, timing);
}

void xg_gemm(const void* A, const void* B, const void* bias, void* C, int M,
             int KP, hipStream_t stream) {
    dim3 grid(M / (xggemm::WAVES * xggemm::WM), xggemm::XN / xggemm::BN);
    dim3 block(xggemm::WAVES * 64);
    if (KP == 512)
        hipLaunchKernelGGL(xggemm::xg_gemm_kernel<512>, grid, block, 0,
                           stream, static_cast<const bf16*>(A),
                           static_cast<const bf16*>(B),
                           static_cast<const bf16*>(bias),
                           static_cast<bf16*>(C), M);
    else if (KP == 256)
        hipLaunchKernelGGL(xggemm::xg_gemm_kernel<256>, grid, block, 0,
                           stream, static_cast<const bf16*>(A),
                           static_cast<const bf16*>(B),
                           static_cast<const bf16*>(bias),
                           static_cast<bf16*>(C), M);
}

int atb_splitk_nslices(int K) {
    return (K + gemmatb::KSLICE - 1) / gemmatb::KSLICE;
}

void atb_splitk_ld(const void* A, int lda, const void* B, int ldb,
                   float* ws, float* C, int M, int N, int K,
                   hipStream_t stream) {
    const int S = atb_splitk_nslices(K);
    dim3 grid(S, (M + gemmatb::BM - 1) / gemmatb::BM,
              (N + gemmatb::BN - 1) / gemmatb::BN);
    hipLaunchKernelGGL(gemmatb::atb_splitk_kernel, grid,
                       dim3(gemmatb::WAVES * 64), 0, stream,
                       static_cast<const bf16*>(A), static_cast<const bf16*>(B),
                       ws, M, N, K, lda, ldb);
    const int64_t MN = (int64_t)M * N;
    // ceil over the 4-element stride: floor(MN/4) left the last MN%4
    // outputs unwritten for odd shapes
    hipLaunchKernelGGL(gemmatb::slice_sum_kernel,
                       dim3(((MN + 3) / 4 + 255) / 256), dim3(256), 0, stream,
                       ws, C, S, MN);
}

void atb_splitk(const void* A, const void* B, float* ws, float* C, int M,
                int N, int K, hipStream_t stream) {
    atb_splitk_ld(A, M, B, N, ws, C, M, N, K, stream);
}

// column sums: out[n] = sum_k X[k*ldx + n], bf16 in / fp32 out. Replaces
// the ones-vector hipBLASLt GEMVs in the weight-grad path (each a separate
// aten launch; the whole weight-grad section is now raw kernel calls —
// the aten form cost ~1.3 ms of HOST enqueue per train step, tr5 trace).
namespace colsum {

__global__ __launch_bounds__(256, 4) void colsum_kernel(
    const bf16* __restrict__ X, float* __restrict__ out, int ldx, int K,
    int N, int ksplit) {
    const int n = blockIdx.x * 256 + threadIdx.x;
    if (n >= N) return;
    const int k0 = blockIdx.y * ksplit;
    const int k1 = min(K, k0 + ksplit);
    // 8 independent accumulators with the loads batched per iteration:
    // a single-accumulator runtime loop compiled to load->wait->add per
    // element (~340 us measured); this form keeps 8 loads in flight
    float acc[8] = {0.f};
    int k = k0;
    for (; k + 8 <= k1; k += 8) {
        float v[8];
#pragma unroll
        for (int q = 0; q < 8; ++q)
            v[q] = bf2f(X[(size_t)(k + q) * ldx + n]);
#pragma unroll
        for (int q = 0; q < 8; ++q) acc[q] += v[q];
    }
    for (; k < k1; ++k) acc[0] += bf2f(X[(size_t)k * ldx + n]);
    // plain store of this k-split's partial (no memset, no atomics: a
    // hipMemsetAsync enqueued just before front_bwd stretched to 659 us
    // of CU starvation — tr6 trace)
    out[(size_t)blockIdx.y * N + n] = ((acc[0] + acc[1]) + (acc[2] + acc[3])) +
                                      ((acc[4] + acc[5]) + (acc[6] + acc[7]));
}

__global__ __launch_bounds__(256, 4) void colsum_reduce_kernel(
    const float* __restrict__ part, float* __restrict__ out, int N,
    int ksp) {
    const int n = blockIdx.x * 256 + threadIdx.x;
    if (n >= N) return;
    float acc = 0.f;
    for (int q = 0; q < ksp; ++q) acc += part[(size_t)q * N + n];
    out[n] = acc;
}

}  // namespace colsum

void colsum_f32(const void* X, int ldx, int K, int N, float* out,
                float* part, hipStream_t stream) {
    const int KSP = 32;
    const int ksplit = (K + KSP - 1) / KSP;
    hipLaunchKernelGGL(colsum::colsum_kernel,
                       dim3((N + 255) / 256, KSP), dim3(256), 0, stream,
                       static_cast<const bf16*>(X), part, ldx, K, N, ksplit);
    hipLaunchKernelGGL(colsum::colsum_reduce_kernel, dim3((N + 255) / 256),
                       dim3(256), 0, stream, part, out, N, KSP);
}

}  // namespace rk
