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
