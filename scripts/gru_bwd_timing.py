"""Per-phase bisection of gru_layer_bwd via its dbg mask."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from roko_amd import ops
ext = ops.ext()
torch.manual_seed(0)
T, B = 90, 128
cache = (torch.randn(T, B, 2, 512, device="cuda") * 0.3).to(torch.bfloat16)
hseq = (torch.randn(T, B, 2, 128, device="cuda") * 0.3).to(torch.bfloat16)
dhin = (torch.randn(T, B, 2, 128, device="cuda") * 0.3).to(torch.bfloat16)
ut = (torch.randn(2, 128, 384, device="cuda") * 0.2).to(torch.bfloat16)

def run(dbg, label, iters=30):
    for _ in range(3):
        ext.gru_layer_bwd(cache, hseq, dhin, ut, dbg)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ext.gru_layer_bwd(cache, hseq, dhin, ut, dbg)
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    print(f"{label:34s} {us:8.1f} us  ({us/T*1000:6.0f} ns/step)")

run(0,  "full")
run(1,  "no global stores")
run(2,  "no gate VALU")
run(4,  "no MFMA")
run(8,  "no staging")
run(15, "shell (barriers only)")
