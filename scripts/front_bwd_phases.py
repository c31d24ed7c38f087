"""Timing experiment: per-phase cost of front_bwd via its phase_mask switch.

Run on a GPU box:  python scripts/front_bwd_phases.py
"""

import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from roko_amd import ops

ext = ops.ext()
torch.manual_seed(0)
B = 128
ids = torch.randint(0, 12, (B, 200, 90), dtype=torch.uint8, device="cuda")
dseq = torch.randn(90, B, 500, device="cuda").to(torch.bfloat16)
w1 = torch.randn(100, 200, device="cuda").to(torch.bfloat16) * 0.1
b1 = torch.randn(100, device="cuda")
w2 = torch.randn(10, 100, device="cuda").to(torch.bfloat16) * 0.1
b2 = torch.randn(10, device="cuda")
emb = torch.randn(12, 50, device="cuda").to(torch.bfloat16)


def timeit(mask, iters=20):
    for _ in range(3):
        ext.front_bwd(ids, dseq, w1, b1, w2, b2, emb, 1234, 0.8, mask)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ext.front_bwd(ids, dseq, w1, b1, w2, b2, emb, 1234, 0.8, mask)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


cases = [
    ("all (0x1F)", 0x1F),
    ("none (loop shell + staging + m build)", 0x00),
    ("G1 only", 0x01),
    ("G1+G3", 0x03),
    ("G1+G3+dt1", 0x07),
    ("G1+G3+dt1+dW (no dm)", 0x0F),
    ("all but G1", 0x1E),
    ("dm only", 0x10),
    ("dW only", 0x08),
]
base = None
for name, m in cases:
    ms = timeit(m)
    print(f"{name:42s} mask=0x{m:02x}  {ms:8.3f} ms")
