"""Phase-level timing breakdown of the inference forward (GPU)."""

import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from roko_amd import config as C
from roko_amd.model import RokoModel
from roko_amd import ops
from roko_amd.ops.forward import _bf16_weights

def main():
    ops.require()
    ext = ops.ext()
    torch.manual_seed(0)
    m = RokoModel().cuda().eval()
    w = _bf16_weights(m)
    B = 128
    T = 90
    x = torch.randint(0, 12, (B, 200, 90), dtype=torch.uint8, device="cuda")

    def timeit(fn, n=50):
        for _ in range(5):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n * 1e6  # us

    seq0 = ext.embed_mlp_fwd(x, w["w1"], w["b1"], w["w2"], w["b2"], w["emb"])
    print(f"embed_mlp: {timeit(lambda: ext.embed_mlp_fwd(x, w['w1'], w['b1'], w['w2'], w['b2'], w['emb'])):8.1f} us")

    xg0 = torch.addmm(w["b_ih0"], seq0.reshape(T * B, -1), w["w_ih_t0"]).view(T, B, 2, 384).contiguous()
    print(f"addmm l0 : {timeit(lambda: torch.addmm(w['b_ih0'], seq0.reshape(T*B,-1), w['w_ih_t0'])):8.1f} us")
    print(f"gru l0   : {timeit(lambda: ext.gru_layer_fwd(xg0, w['u0'], w['bhh0'], False)):8.1f} us")

    h0 = ext.gru_layer_fwd(xg0, w["u0"], w["bhh0"], False)[0].view(T, B, 256)
    xg1 = torch.addmm(w["b_ih1"], h0.reshape(T * B, -1), w["w_ih_t1"]).view(T, B, 2, 384).contiguous()
    print(f"addmm l1 : {timeit(lambda: torch.addmm(w['b_ih1'], h0.reshape(T*B,-1), w['w_ih_t1'])):8.1f} us")
    print(f"gru l1   : {timeit(lambda: ext.gru_layer_fwd(xg1, w['u1'], w['bhh1'], False)):8.1f} us")
    print(f"head     : {timeit(lambda: ext.head_fwd(h0.contiguous(), w['w4'], w['b4'], False, True)):8.1f} us")

    from roko_amd.ops.forward import roko_argmax
    print(f"full fwd (argmax, no D2H): {timeit(lambda: roko_argmax(m, x)):8.1f} us")
    print(f"full fwd + D2H           : {timeit(lambda: roko_argmax(m, x).cpu()):8.1f} us")

    # multi-batch concurrency: does overlapping batches on streams help?
    xs = [torch.randint(0, 12, (B, 200, 90), dtype=torch.uint8, device="cuda") for _ in range(8)]
    streams = [torch.cuda.Stream() for _ in range(8)]
    for nconc in (1, 2, 4, 8):
        def multi():
            for i in range(nconc):
                with torch.cuda.stream(streams[i]):
                    roko_argmax(m, xs[i])
            torch.cuda.synchronize()
        us = timeit(multi, n=20)
        print(f"{nconc} concurrent batches: {us:8.1f} us total, {us/nconc:8.1f} us/batch")

if __name__ == "__main__":
    import sys as _s0
    if not (len(_s0.argv) > 1 and _s0.argv[1] == "train"):
        main()

def train_breakdown():
    import torch
    from roko_amd.model import RokoModel
    from roko_amd.ops.train import FusedAdam, fused_train_step, train_forward, fused_cross_entropy
    torch.manual_seed(0)
    m = RokoModel().cuda().train()
    opt = FusedAdam(list(m.parameters()))
    x = torch.randint(0, 12, (128, 200, 90), dtype=torch.uint8, device="cuda")
    y = torch.randint(0, 5, (128, 90), device="cuda")
    import time
    def timeit(fn, n=20):
        for _ in range(5): fn()
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(n): fn()
        torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e6

    print(f"train fwd            : {timeit(lambda: train_forward(m, x)):9.1f} us")
    def fwd_loss():
        return fused_cross_entropy(train_forward(m, x), y)
    print(f"train fwd+loss       : {timeit(fwd_loss):9.1f} us")
    def full():
        fused_train_step(m, x, y, opt)
    print(f"full step            : {timeit(full):9.1f} us")
    # torch profiler table
    from torch.profiler import profile, ProfilerActivity
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
        for _ in range(3):
            fused_train_step(m, x, y, opt)
        torch.cuda.synchronize()
    print(prof.key_averages().table(sort_by="cuda_time_total", row_limit=18))

if __name__ == "__main__":
    import sys as _s
    if len(_s.argv) > 1 and _s.argv[1] == "train":
        train_breakdown()
