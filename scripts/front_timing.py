"""A/B the eval front kernels (v1 per-column, v2 chunked, v3 wave-private)
and correctness-vs-v2 on random ids."""
import os
import sys
import time

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

o2 = ext.embed_mlp_fwd2(x, w["w1g"], w["b1"], w["w2"], w["b2"], w["emb"])
o3 = ext.embed_mlp_fwd3(x, w["w1g"], w["b1"], w["w2"], w["b2"], w["emb"])
d = (o2.float() - o3.float()).abs()
print(f"v3 vs v2: max |d| {d.max().item():.5f} mean {d.mean().item():.6f}")


def bench(label, fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    print(f"{label:24s} {us:8.1f} us/kernel")


bench("v1 per-column", lambda: ext.embed_mlp_fwd(
    x, w["w1"], w["b1"], w["w2"], w["b2"], w["emb"]))
bench("v2 chunked", lambda: ext.embed_mlp_fwd2(
    x, w["w1g"], w["b1"], w["w2"], w["b2"], w["emb"]))
bench("v3 wave-private", lambda: ext.embed_mlp_fwd3(
    x, w["w1g"], w["b1"], w["w2"], w["b2"], w["emb"]))
