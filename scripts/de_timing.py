import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from roko_amd import ops
ext = ops.ext()
torch.manual_seed(0)
B = 128
ids = torch.randint(0, 12, (B, 200, 90), dtype=torch.uint8, device="cuda")
dt1g = (torch.randn(B, 90, 112, 64, device="cuda") * 0.1).to(torch.bfloat16)
w1 = (torch.randn(100, 200, device="cuda") * 0.1).to(torch.bfloat16)

def run(dbg, label):
    for _ in range(3):
        ext.front_de_timed(ids, dt1g, w1, 1234, 0.8, dbg)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 20
    tims = torch.zeros(4, dtype=torch.int64)
    for _ in range(iters):
        de, tim = ext.front_de_timed(ids, dt1g, w1, 1234, 0.8, dbg)
        tims += tim.cpu()
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / iters * 1e3
    per = tims.double() / iters / (B * 2)
    print(f"{label:28s} wall {ms:7.3f} ms  per-col: stage={per[0]/45:6.0f} gemm+epi={per[1]/45:6.0f} cyc")

run(0, "full")
run(1, "no-atomics (reg sink)")
run(2, "no-hash")
run(3, "no-atomics no-hash")
run(4, "no-epilogue (MFMA+loads)")
