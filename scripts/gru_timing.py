import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from roko_amd import ops
ext = ops.ext()
torch.manual_seed(0)
T, B = 90, 128
xg = (torch.randn(T, B, 2, 384, device="cuda") * 0.3).to(torch.bfloat16)
u = (torch.randn(2, 384, 128, device="cuda") * 0.2).to(torch.bfloat16)
bhh = torch.randn(2, 384, device="cuda")

def run(train, dbg, label, iters=30):
    for _ in range(3):
        ext.gru_layer_fwd(xg, u, bhh, train, dbg)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ext.gru_layer_fwd(xg, u, bhh, train, dbg)
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    print(f"{label:38s} {us:8.1f} us  ({us/T*1000:6.0f} ns/step)")

for train in (False, True):
    tag = "train" if train else "eval "
    run(train, 0,  f"[{tag}] full")
    run(train, 1,  f"[{tag}] no hseq store")
    run(train, 3,  f"[{tag}] no stores")
    run(train, 4,  f"[{tag}] no gate VALU")
    run(train, 8,  f"[{tag}] no xg staging")
    run(train, 16, f"[{tag}] no MFMA")
    run(train, 31, f"[{tag}] shell (barriers+LDS h only)")

# fused (in-kernel xg GEMM) vs split timing per layer shape
from roko_amd.model import RokoModel
from roko_amd.ops import forward as fwd
m = RokoModel().cuda().eval()
w = fwd._bf16_weights(m)
for l, IN in ((0, 500), (1, 256), (2, 256)):
    x = (torch.randn(T, B, IN, device="cuda") * 0.3).to(torch.bfloat16)
    def split():
        xg2 = torch.addmm(w[f"b_ih{l}"], x.reshape(T * B, -1), w[f"w_ih_t{l}"]).view(T, B, 2, 384)
        ext.gru_layer_fwd(xg2.contiguous(), w[f"u{l}"], w[f"bhh{l}"], False)
    def fused():
        ext.gru_layer_fused(x, w[f"w_ih_p{l}"], w[f"b_ih{l}"], w[f"u{l}"], w[f"bhh{l}"])
    for name, fn in (("split", split), ("fused", fused)):
        for _ in range(3): fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(20): fn()
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / 20 * 1e6
        print(f"layer{l} IN={IN} {name:6s} {us:8.1f} us")
