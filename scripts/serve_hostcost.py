"""Host-vs-GPU attribution for the serving hot loop.

Submits N batches without waiting, then times the queue drain:
  - submit-loop wall >> drain  => host-submit-bound
  - drain >> 0                 => GPU-bound (queue backed up)
Also times the raw pybind submit with an empty GPU (first call after sync).
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from roko_amd import ops
from roko_amd.model import RokoModel
from roko_amd.ops.forward import InferencePipeline

batch = int(sys.argv[1]) if len(sys.argv) > 1 else 128
N = 4000
torch.manual_seed(0)
model = RokoModel().cuda().eval()
ops.require()
pipe = InferencePipeline(model, batch, depth=48)
x = torch.randint(0, 12, (batch, 200, 90), dtype=torch.uint8).cuda()
for _ in range(100):
    pipe.submit(x, copy_out=False)
torch.cuda.synchronize()

t0 = time.perf_counter()
for _ in range(N):
    pipe.submit(x, copy_out=False)
t1 = time.perf_counter()
torch.cuda.synchronize()
t2 = time.perf_counter()
sub_us = (t1 - t0) / N * 1e6
drain = t2 - t1
print(f"batch {batch}: submit loop {sub_us:.1f} us/batch, drain {drain*1e3:.0f} ms "
      f"({drain/(t2-t0)*100:.0f}% of total)")
print(f"implied host-capped rate: {batch*30/sub_us:.2f}M bases/s; "
      f"measured overall: {N*batch*30/(t2-t0)/1e6:.2f}M")
