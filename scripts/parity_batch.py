"""Pipeline-vs-eager parity at large serving batches."""
import sys, torch
sys.path.insert(0, ".")
from roko_amd.model import RokoModel
from roko_amd import ops
from roko_amd.ops.forward import InferencePipeline, roko_argmax

torch.manual_seed(0)
model = RokoModel().cuda().eval()
ops.require()
for b in (256, 512, 1024):
    pipe = InferencePipeline(model, b, depth=4)
    g = torch.Generator().manual_seed(11)
    x = torch.randint(0, 12, (b, 200, 90), generator=g, dtype=torch.uint8).cuda()
    t = pipe.submit(x)
    got = t()
    ref = roko_argmax(model, x).cpu()
    eq = (got == ref).float().mean().item()
    print(f"b={b}: match {eq:.6f}")
    assert eq > 0.999, eq
    del pipe
print("PARITY_OK")
