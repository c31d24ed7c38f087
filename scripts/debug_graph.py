"""Diagnose the graphed-train-step learning stall after warmup-state restore."""
import torch

from roko_amd.model import RokoModel
from roko_amd.ops.train import FusedAdam, GraphedTrainStep, fused_train_step

torch.manual_seed(13)
m = RokoModel().cuda().train()
opt = FusedAdam(list(m.parameters()), lr=3e-3)
p0 = opt.flat_p.clone()
step = GraphedTrainStep(m, opt, batch=32)
torch.cuda.synchronize()
print("after init: |dp|max", (opt.flat_p - p0).abs().max().item(),
      "step_buf", int(step.step_buf.item()),
      "|m|max", opt.m.abs().max().item())
x = torch.randint(0, 12, (32, 200, 90), dtype=torch.uint8, device="cuda")
y = torch.randint(0, 5, (32, 90), device="cuda")
losses = []
for i in range(80):
    l = float(step(x, y))
    losses.append(l)
    if i < 3 or i % 10 == 0:
        torch.cuda.synchronize()
        print(f"{i}: loss {l:.4f} |dp| {(opt.flat_p - p0).abs().max().item():.5f}"
              f" |g| {opt.flat_g.abs().max().item():.5f}"
              f" |m| {opt.m.abs().max().item():.5f}"
              f" step_buf {int(step.step_buf.item())}")
print("graphed: first", losses[0], "last", losses[-1])

torch.manual_seed(13)
m2 = RokoModel().cuda().train()
o2 = FusedAdam(list(m2.parameters()), lr=3e-3)
ls = [float(fused_train_step(m2, x, y, o2)) for _ in range(80)]
print("eager:   first", ls[0], "last", ls[-1])
