"""Numerics + timing for xg_gemm2 (the LDS-staged specialized xg GEMM)
vs the hipBLASLt addmm it would replace in serving.

Usage (GPU): python scripts/xg2_timing.py
"""
import torch

import roko_amd  # noqa: F401  (env defaults)
from roko_amd.ops import _hip_ops as ext


def pad_w(w, kp):
    out = torch.zeros(768, kp, dtype=torch.bfloat16, device="cuda")
    out[:, : w.shape[1]] = w
    return out.contiguous()


def bench(fn, iters=200):
    s = torch.cuda.Stream()
    with torch.cuda.stream(s):
        for _ in range(20):
            fn()
        torch.cuda.synchronize()
        e0 = torch.cuda.Event(enable_timing=True)
        e1 = torch.cuda.Event(enable_timing=True)
        e0.record()
        for _ in range(iters):
            fn()
        e1.record()
        torch.cuda.synchronize()
    return e0.elapsed_time(e1) / iters * 1e3  # us


def main():
    torch.manual_seed(0)
    B = 128
    M = 90 * B
    for kreal, kp, tag in [(500, 512, "l0"), (256, 256, "l1/l2")]:
        A = torch.randn(M, kreal, device="cuda").bfloat16().contiguous()
        W = torch.randn(768, kreal, device="cuda").bfloat16() * 0.05
        bias = (torch.randn(768, device="cuda") * 0.1).bfloat16()
        Wp = pad_w(W, kp)
        Wt = W.t().contiguous()

        ref = torch.addmm(bias.float(), A.float(), W.t().float())
        out = ext.xg_gemm2(A, Wp, bias)
        err = (out.float() - ref).abs().max().item()
        rel = err / ref.abs().max().item()
        # bf16 accumulate in fp32 → tight tolerance vs fp32 ref
        ok = rel < 2e-2
        print(f"{tag}: max abs err {err:.4f} rel {rel:.2e} {'OK' if ok else 'FAIL'}")

        t2 = bench(lambda: ext.xg_gemm2(A, Wp, bias))
        tim = torch.zeros(4, dtype=torch.int64, device="cuda")
        ext.xg_gemm2(A, Wp, bias, tim)
        torch.cuda.synchronize()
        ph = tim.cpu().tolist()
        print(f"{tag}: WG0 phase cycles commit+issue={ph[0]} mfma={ph[1]} "
              f"barrier={ph[2]} epilogue={ph[3]} total={sum(ph)}")
        xgbuf = torch.empty(M, 768, device="cuda", dtype=torch.bfloat16)
        t1 = bench(lambda: torch.addmm(bias, A, Wt, out=xgbuf))
        flops = 2 * M * 768 * kreal
        print(f"{tag}: xg_gemm2 {t2:.1f} us ({flops / t2 / 1e6:.0f} GF/s)  "
              f"addmm {t1:.1f} us ({flops / t1 / 1e6:.0f} GF/s)  "
              f"speedup {t1 / t2:.2f}x")


if __name__ == "__main__":
    main()
