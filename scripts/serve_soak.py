"""Serving soak: sustained pipelined inference for N seconds; asserts
throughput steadiness and flat device-memory use (no leak per batch)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from roko_amd import config as C
from roko_amd import ops
from roko_amd.model import RokoModel
from roko_amd.ops.forward import InferencePipeline

secs = float(sys.argv[1]) if len(sys.argv) > 1 else 60.0
batch = int(sys.argv[2]) if len(sys.argv) > 2 else 128
depth = int(sys.argv[3]) if len(sys.argv) > 3 else 48
torch.manual_seed(0)
model = RokoModel().cuda().eval()
ops.require()
pipe = InferencePipeline(model, batch, depth=depth)
g = torch.Generator().manual_seed(7)
xs = [torch.randint(0, 12, (batch, 200, 90), generator=g,
                    dtype=torch.uint8).cuda()
      for _ in range(64)]
for i in range(100):
    pipe.submit(xs[i % 64], copy_out=False)
torch.cuda.synchronize()
mem0 = torch.cuda.memory_allocated()
windows = 0
t0 = time.perf_counter()
marks = []
while True:
    for _ in range(max(256000 // batch, 200)):
        pipe.submit(xs[windows % 64], copy_out=False)
        windows += batch
    torch.cuda.synchronize()
    el = time.perf_counter() - t0
    marks.append(windows * 30 / el)
    if el >= secs:
        break
mem1 = torch.cuda.memory_allocated()
print(f"soak {el:.1f}s: {windows} windows, {windows*30/el/1e6:.2f}M bases/s")
print(f"throughput marks (M bases/s): {[round(m/1e6,2) for m in marks[:3]]}"
      f" ... {[round(m/1e6,2) for m in marks[-3:]]}")
print(f"device mem: {mem0/1e6:.1f} -> {mem1/1e6:.1f} MB (delta {mem1-mem0} B)")
assert mem1 == mem0, "device memory grew during serving"
first, last = marks[0], marks[-1]
assert last > 0.9 * first, (first, last)
print("SOAK_OK")
