"""Per-phase timing of the v3 eval front via its timing buffer."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from roko_amd import ops
from roko_amd.model import RokoModel
from roko_amd.ops import forward as fwd

ext = ops.ext()
torch.manual_seed(0)
m = RokoModel().cuda().eval()
w = fwd._bf16_weights(m)
B = 128
x = torch.randint(0, 12, (B, 200, 90), dtype=torch.uint8, device="cuda")
tim = torch.zeros(6, dtype=torch.int64, device="cuda")
iters = 30
for _ in range(3):
    ext.embed_mlp_fwd3(x, w["w1g"], w["b1"], w["w2"], w["b2"], w["emb"])
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(iters):
    ext.embed_mlp_fwd3(x, w["w1g"], w["b1"], w["w2"], w["b2"], w["emb"], tim)
torch.cuda.synchronize()
us = (time.perf_counter() - t0) / iters * 1e6
t = tim.cpu().numpy() / iters  # cycles per kernel (lane 0, wave 0: 23 cols)
names = ["scatter", "G1", "G2", "G3", "store"]
tot = t[:5].sum()
print(f"kernel {us:.1f} us (timed run incl. instrumentation)")
for n, v in zip(names, t[:5]):
    print(f"  {n:8s} {v:12.0f} cyc  {100*v/tot:5.1f}%   {v/23:8.0f} cyc/col")
