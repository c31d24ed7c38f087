"""Host-vs-GPU attribution for the fused train step: submit N steps
without syncing, then time the drain. Submit-loop >> drain => host-bound."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from roko_amd import config as C
from roko_amd.model import RokoModel
from roko_amd.ops.train import FusedAdam, fused_param_order, fused_train_step

N = 60
torch.manual_seed(0)
model = RokoModel().cuda().train()
opt = FusedAdam(fused_param_order(model), lr=C.LR)
g = torch.Generator().manual_seed(7)
x = torch.randint(0, C.NUM_BASE_IDS, (128, C.WINDOW_ROWS, C.WINDOW_COLS),
                  generator=g, dtype=torch.uint8).cuda()
y = torch.randint(0, C.NUM_CLASSES, (128, C.WINDOW_COLS), generator=g).cuda()
for _ in range(10):
    fused_train_step(model, x, y, opt)
torch.cuda.synchronize()

t0 = time.perf_counter()
for _ in range(N):
    fused_train_step(model, x, y, opt)
t1 = time.perf_counter()
torch.cuda.synchronize()
t2 = time.perf_counter()
sub = (t1 - t0) / N * 1e3
tot = (t2 - t0) / N * 1e3
print(f"submit {sub:.2f} ms/step, drain {(t2-t1)*1e3:.0f} ms "
      f"({(t2-t1)/(t2-t0)*100:.0f}% of total), overall {tot:.2f} ms/step")
