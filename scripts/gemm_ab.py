"""A/B hipBLASLt (torch.addmm/mm) vs the in-tree gemm_bias kernel on the
train/serving hot GEMM shapes (xg projections and dx)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from roko_amd import ops

ext = ops.ext()
torch.manual_seed(0)
TB = 90 * 128

shapes = [
    ("xg l0 (TB,500)x(500,768)", TB, 768, 500),
    ("xg l12 (TB,256)x(256,768)", TB, 768, 256),
    ("dx l0 (TB,768)x(768,500)", TB, 500, 768),
    ("dx l12 (TB,768)x(768,256)", TB, 256, 768),
]


def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


for label, M, N, K in shapes:
    A = (torch.randn(M, K, device="cuda") * 0.3).to(torch.bfloat16)
    Bm = (torch.randn(K, N, device="cuda") * 0.3).to(torch.bfloat16)
    bias = torch.randn(N, device="cuda", dtype=torch.bfloat16)
    out = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
    t_lt = bench(lambda: torch.addmm(bias, A, Bm, out=out))
    t_mm = bench(lambda: torch.mm(A, Bm, out=out))
    t_ours = bench(lambda: ext.gemm_bias(A, Bm, bias.float()))
    # correctness
    ref = torch.addmm(bias.float(), A.float(), Bm.float())
    got = ext.gemm_bias(A, Bm, bias.float()).float()
    rel = (got - ref).norm() / ref.norm()
    print(f"{label:28s} addmm {t_lt:7.1f}  mm {t_mm:7.1f}  "
          f"gemm_bias {t_ours:7.1f} us  rel {rel.item():.4f}")

# train-path layout question: addmm with a TRANSPOSED-VIEW B picks hipBLASLt
# MT16x16x256 (24 us) while the contiguous layout gets MT256x160 (16 us)
print("\n-- B-operand layout (train xg GEMM shapes) --")
for label, M, N, K in shapes[:2]:
    A = (torch.randn(M, K, device="cuda") * 0.3).to(torch.bfloat16)
    W = (torch.randn(N, K, device="cuda") * 0.3).to(torch.bfloat16)  # (768,K)
    bias = torch.randn(N, device="cuda", dtype=torch.bfloat16)
    out = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
    t_view = bench(lambda: torch.addmm(bias, A, W.t(), out=out))
    def with_copy():
        Wt = W.t().contiguous()
        torch.addmm(bias, A, Wt, out=out)
    t_copy = bench(with_copy)
    print(f"{label:28s} B=view {t_view:7.1f}  B=copy(+transpose) {t_copy:7.1f} us")

# custom LDS-free xg_gemm vs hipBLASLt at the padded serving shapes
print("\n-- xg_gemm (custom) --")
for KP, Kr in ((512, 500), (256, 256)):
    M = 90 * 128
    A = torch.zeros(M, KP, device="cuda", dtype=torch.bfloat16)
    A[:, :Kr] = (torch.randn(M, Kr, device="cuda") * 0.3).to(torch.bfloat16)
    Bt = (torch.randn(768, KP, device="cuda") * 0.3).to(torch.bfloat16)
    Bt[:, Kr:] = 0
    bias = (torch.randn(768, device="cuda") * 0.3).to(torch.bfloat16)
    out = torch.empty(M, 768, device="cuda", dtype=torch.bfloat16)
    t_lt = bench(lambda: torch.addmm(bias, A, Bt.t(), out=out))
    t_x = bench(lambda: ext.xg_gemm(A, Bt, bias))
    ref = torch.addmm(bias.float(), A.float(), Bt.float().t())
    got = ext.xg_gemm(A, Bt, bias).float()
    rel = (got - ref).norm() / ref.norm()
    gf = 2.0 * M * 768 * Kr
    print(f"KP={KP}: addmm {t_lt:6.1f} us ({gf/t_lt/1e6:.0f} TF/s)   "
          f"xg_gemm {t_x:6.1f} us ({gf/t_x/1e6:.0f} TF/s)  rel {rel:.4f}")
