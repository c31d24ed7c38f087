// Persistent bidirectional GRU layer forward for gfx950.
//
// Replaces the reference's cuDNN GRU (rnn_model.py:57).
// Design (SURVEY.md §7 hard part (a), re-thought for CDNA4):
//   * the input projections W_ih·x + b_ih for ALL T steps are one plain GEMM
//     done on the host side (hipBLASLt via torch.matmul) — this kernel gets
//     the precomputed per-step gate inputs `xg`;
//   * ONE kernel launch runs the full T-step recurrence of a layer: grid =
//     (B/32 batch tiles) x (2 directions); each workgroup owns 32 batch rows
//     whose hidden state lives in fp32 registers (fragment-shaped) with a
//     bf16 mirror in LDS as the MFMA A-operand — no per-step launches, no
//     grid-wide sync (batch rows are independent);
//   * U = weight_hh^T fragments are loaded ONCE into registers (12 bf16x8
//     per wave) and reused for all T steps; per-step work is 24
//     v_mfma_f32_16x16x32_bf16 per wave + fused sigmoid/tanh gate math;
//   * per-step xg tiles (32x384 bf16) are double-buffered through LDS so the
//     HBM reads of step t+1 overlap the MFMAs of step t.
//
// PyTorch GRU semantics (gate order r,z,n in weight rows; n-gate bias split):
//   r = sigmoid(xr + U_r h + bhh_r)
//   z = sigmoid(xz + U_z h + bhh_z)
//   n = tanh(xn + r * (U_n h + bhh_n))        (xg already contains b_ih)
//   h' = (1 - z) * n + z * h

#include <cstdint>

#include "common.h"

namespace rk {

constexpr int H = 128;       // hidden size (config.HIDDEN_SIZE)
constexpr int G3 = 3 * H;    // gate rows
constexpr int MB = 32;       // batch rows per workgroup
constexpr int WAVES = 8;     // 512 threads
constexpr int HPAD = H + 8;  // LDS row padding (bank-conflict fix)

__global__ __launch_bounds__(WAVES * 64) void gru_layer_fwd_kernel(
    const bf16* __restrict__ xg,   // (T, B, 2, 3H)  W_ih·x + b_ih
    const bf16* __restrict__ u,    // (2, 3H, H)     weight_hh
    const float* __restrict__ bhh, // (2, 3H)        bias_hh
    bf16* __restrict__ hseq,       // (T, B, 2, H)   output
    int T, int B) {
    __shared__ struct {
        bf16 h[MB][HPAD];            // bf16 mirror of the hidden state
        bf16 xgb[2][MB][G3];         // double-buffered step gate inputs
    } lds;

    const int dir = blockIdx.y;
    const int b0 = blockIdx.x * MB;
    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int j0 = wid * 16;         // this wave's 16 hidden columns
    const int lrow = lane >> 4;      // fragment row group (0..3)
    const int lcol = lane & 15;      // fragment column

    // ---- load U fragments (kept in registers for all T steps) -------------
    // B-fragment for gates = h·U^T: B[k][col] = U[gate*H + j0 + col][k]
    bf16x8 ufrag[3][4];
#pragma unroll
    for (int g = 0; g < 3; ++g) {
#pragma unroll
        for (int kb = 0; kb < 4; ++kb) {
            const int jrow = g * H + j0 + lcol;
            const int k = kb * 32 + 8 * lrow;
            ufrag[g][kb] = *reinterpret_cast<const bf16x8*>(
                u + (size_t)dir * G3 * H + (size_t)jrow * H + k);
        }
    }
    float bhh_reg[3];
#pragma unroll
    for (int g = 0; g < 3; ++g) bhh_reg[g] = bhh[dir * G3 + g * H + j0 + lcol];

    // ---- zero hidden state -------------------------------------------------
    float hreg[2][4];  // fp32 master copy, fragment-shaped (rows of this wave)
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
        for (int i = 0; i < 4; ++i) hreg[mt][i] = 0.0f;
    for (int e = tid; e < MB * HPAD; e += WAVES * 64) lds.h[0][e] = f2bf(0.0f);

    // ---- stage xg for the first step --------------------------------------
    const int t_first = (dir == 0) ? 0 : T - 1;
    {
        const bf16* src = xg + (((size_t)t_first * B + b0) * 2 + dir) * G3;
#pragma unroll
        for (int p = 0; p < 3; ++p) {
            int e = (p * WAVES * 64 + tid) * 8;
            int row = e / G3, col = e % G3;
            *reinterpret_cast<bf16x8*>(&lds.xgb[0][row][col]) =
                *reinterpret_cast<const bf16x8*>(src + (size_t)row * 2 * G3 + col);
        }
    }
    __syncthreads();

    // ---- T-step recurrence -------------------------------------------------
    int cur = 0;
    for (int ti = 0; ti < T; ++ti) {
        const int t = (dir == 0) ? ti : T - 1 - ti;
        // stage next step's xg into the other buffer
        if (ti + 1 < T) {
            const int tn = (dir == 0) ? ti + 1 : T - 2 - ti;
            const bf16* src = xg + (((size_t)tn * B + b0) * 2 + dir) * G3;
#pragma unroll
            for (int p = 0; p < 3; ++p) {
                int e = (p * WAVES * 64 + tid) * 8;
                int row = e / G3, col = e % G3;
                *reinterpret_cast<bf16x8*>(&lds.xgb[cur ^ 1][row][col]) =
                    *reinterpret_cast<const bf16x8*>(src + (size_t)row * 2 * G3 + col);
            }
        }

        // gates_h = h · U^T  (24 MFMA per wave)
        f32x4 acc[2][3];
#pragma unroll
        for (int mt = 0; mt < 2; ++mt)
#pragma unroll
            for (int g = 0; g < 3; ++g) acc[mt][g] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kb = 0; kb < 4; ++kb) {
#pragma unroll
            for (int mt = 0; mt < 2; ++mt) {
                bf16x8 a = lds_load_a_frag(&lds.h[0][0], mt * 16, kb * 32, HPAD);
#pragma unroll
                for (int g = 0; g < 3; ++g)
                    acc[mt][g] = mfma16x16x32(a, ufrag[g][kb], acc[mt][g]);
            }
        }

        // fused gate math; updates the fp32 register hidden state
#pragma unroll
        for (int mt = 0; mt < 2; ++mt) {
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int row = mt * 16 + lrow * 4 + i;
                const float xr = bf2f(lds.xgb[cur][row][0 * H + j0 + lcol]);
                const float xz = bf2f(lds.xgb[cur][row][1 * H + j0 + lcol]);
                const float xn = bf2f(lds.xgb[cur][row][2 * H + j0 + lcol]);
                const float r = sigmoidf_dev(xr + acc[mt][0][i] + bhh_reg[0]);
                const float z = sigmoidf_dev(xz + acc[mt][1][i] + bhh_reg[1]);
                const float n = tanhf_dev(xn + r * (acc[mt][2][i] + bhh_reg[2]));
                hreg[mt][i] = (1.0f - z) * n + z * hreg[mt][i];
            }
        }

        __syncthreads();  // all waves done reading lds.h (and prior hseq read)
#pragma unroll
        for (int mt = 0; mt < 2; ++mt)
#pragma unroll
            for (int i = 0; i < 4; ++i)
                lds.h[mt * 16 + lrow * 4 + i][j0 + lcol] = f2bf(hreg[mt][i]);
        __syncthreads();  // new h visible

        // cooperative wide store of h to hseq (coalesced 16B per lane)
        {
            bf16* dst = hseq + (((size_t)t * B + b0) * 2 + dir) * H;
            const int row = tid / 16;           // 512 threads = 32 rows x 16
            const int col = (tid % 16) * 8;
            *reinterpret_cast<bf16x8*>(dst + (size_t)row * 2 * H + col) =
                *reinterpret_cast<const bf16x8*>(&lds.h[row][col]);
        }
        cur ^= 1;
    }
}

void gru_layer_fwd(const void* xg, const void* u, const float* bhh, void* hseq,
                   int T, int B, hipStream_t stream) {
    dim3 grid(B / MB, 2);
    dim3 block(WAVES * 64);
    hipLaunchKernelGGL(gru_layer_fwd_kernel, grid, block, 0, stream,
                       static_cast<const bf16*>(xg), static_cast<const bf16*>(u),
                       bhh, static_cast<bf16*>(hseq), T, B);
}

}  // namespace rk
