"""Numerics check: ext.gru_wgrads / ext.head_wgrads vs the aten reference."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from roko_amd.ops import _hip_ops as ext

torch.manual_seed(0)
T, B, H, IN = 90, 128, 128, 500
TB, G3 = T * B, 3 * H
dev = "cuda"
dhg = (torch.randn(2, T, B, G3, device=dev) * 0.1).bfloat16().contiguous()
dxg = (torch.randn(TB, 2 * G3, device=dev) * 0.1).bfloat16().contiguous()
hseq = (torch.randn(T, B, 2, H, device=dev) * 0.5).bfloat16().contiguous()
x = (torch.randn(TB, IN, device=dev) * 0.5).bfloat16().contiguous()

# aten reference (the non-deferred path's math)
zeros = hseq.new_zeros(1, B, H)
hp_f = torch.cat([zeros, hseq[:-1, :, 0, :]], 0).reshape(TB, H)
hp_r = torch.cat([hseq[1:, :, 1, :], zeros], 0).reshape(TB, H)
dhg_f = dhg[0].reshape(TB, G3)
dhg_r = dhg[1].reshape(TB, G3)
du_ref = torch.stack([dhg_f.t().float() @ hp_f.float(),
                      dhg_r.t().float() @ hp_r.float()])
dw_ref = dxg.t().float() @ x.float()
dbhh_ref = torch.stack([dhg_f.float().sum(0), dhg_r.float().sum(0)])
dbih_ref = dxg.float().sum(0)

S = (TB + 255) // 256
ws = torch.empty(S, 2 * G3, IN, device=dev, dtype=torch.float32)
du = torch.empty(2, G3, H, device=dev, dtype=torch.float32)
dw = torch.empty(2 * G3, IN, device=dev, dtype=torch.float32)
dbhh = torch.empty(2, G3, device=dev, dtype=torch.float32)
dbih = torch.empty(2 * G3, device=dev, dtype=torch.float32)
ext.gru_wgrads(dhg, dxg, hseq, x, ws, ws, du, dw, dbhh, dbih)
torch.cuda.synchronize()
for name, got, ref in [("du", du, du_ref), ("dw", dw, dw_ref),
                       ("dbhh", dbhh, dbhh_ref), ("dbih", dbih, dbih_ref)]:
    err = (got - ref).abs().max().item()
    rel = err / max(ref.abs().max().item(), 1e-9)
    print(f"{name}: max abs {err:.4f} rel {rel:.2e}", "OK" if rel < 1e-2 else "FAIL")

dl = (torch.randn(TB, 5, device=dev) * 0.1).bfloat16().contiguous()
seq = hseq.reshape(TB, 2 * H)
dw4 = torch.empty(5, 2 * H, device=dev, dtype=torch.float32)
db4 = torch.empty(5, device=dev, dtype=torch.float32)
ext.head_wgrads(dl, seq, ws, dw4, db4)
torch.cuda.synchronize()
print("dw4 rel", ((dw4 - dl.t().float() @ seq.float()).abs().max() /
                  dw4.abs().max()).item())
print("db4 rel", ((db4 - dl.float().sum(0)).abs().max() /
                  db4.abs().max()).item())
