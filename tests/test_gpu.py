"""GPU kernel numerics tests: every HIP kernel vs the plain PyTorch fp32
reference of the same op (SURVEY.md §4 test strategy (c))."""

import numpy as np
import pytest
import torch

from roko_amd import config as C
from roko_amd.model import RokoModel

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from roko_amd import ops
    from roko_amd.ops import forward as fwd

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs ROCm GPU"
)


@requires_gpu
def test_extension_loaded_and_native():
    # fail LOUDLY if the extension is absent on a GPU host
    ops.require()
    assert ops.available()


@requires_gpu
def test_mfma_fragment_layouts():
    """Asymmetric-operand probe validates the A/B/C lane mappings
    (cdna_hip_programming.md §3 'Always A=I-check with ASYMMETRIC B')."""
    ext = ops.ext()
    torch.manual_seed(0)
    a = torch.randn(16, 32, device="cuda") * 0.5
    b = torch.randn(32, 16, device="cuda") * 0.5
    d = ext.mfma_probe(a, b)
    ref = (a.to(torch.bfloat16).float() @ b.to(torch.bfloat16).float())
    assert torch.allclose(d, ref, atol=2e-2, rtol=2e-2), (
        (d - ref).abs().max().item()
    )


@requires_gpu
def test_embed_mlp_fwd_vs_torch():
    torch.manual_seed(1)
    m = RokoModel().eval()
    B = 4
    x = torch.randint(0, 12, (B, 200, 90))
    with torch.no_grad():
        e = m.embedding(x)
        t = torch.relu(m.fc1(e.permute(0, 2, 3, 1)))
        t = torch.relu(m.fc2(t))
        ref = t.reshape(B, 90, 500).transpose(0, 1)  # (90,B,500)

    m = m.cuda()
    w = fwd._bf16_weights(m)
    ext = ops.ext()
    out = ext.embed_mlp_fwd(
        x.to(torch.uint8).cuda(), w["w1"], w["b1"], w["w2"], w["b2"], w["emb"]
    )
    got = out.float().cpu()
    err = (got - ref).abs()
    scale = ref.abs().mean().item() + 1e-6
    assert err.max().item() < 0.08, err.max().item()
    assert err.mean().item() / scale < 0.02


@requires_gpu
@pytest.mark.parametrize("version", ["v2", "v3"])
def test_embed_mlp_fwd_v2_v3_vs_torch(version):
    """The chunked (v2) and wave-private-column (v3) eval fronts against the
    fp32 torch reference, incl. the padded-batch path."""
    torch.manual_seed(21)
    m = RokoModel().eval()
    B = 32
    x = torch.randint(0, 12, (B, 200, 90))
    with torch.no_grad():
        e = m.embedding(x)
        t = torch.relu(m.fc1(e.permute(0, 2, 3, 1)))
        t = torch.relu(m.fc2(t))
        ref = t.reshape(B, 90, 500).transpose(0, 1)

    m = m.cuda()
    w = fwd._bf16_weights(m)
    ext = ops.ext()
    fn = ext.embed_mlp_fwd2 if version == "v2" else ext.embed_mlp_fwd3
    out = fn(x.to(torch.uint8).cuda(), w["w1g"], w["b1"], w["w2"], w["b2"],
             w["emb"])
    got = out.float().cpu()
    err = (got - ref).abs()
    scale = ref.abs().mean().item() + 1e-6
    assert err.max().item() < 0.08, err.max().item()
    assert err.mean().item() / scale < 0.02


@requires_gpu
def test_gru_layer_fwd_vs_torch():
    torch.manual_seed(2)
    T, B, H = 90, 32, 128
    gru = torch.nn.GRU(256, H, num_layers=1, batch_first=False,
                       bidirectional=True)
    x = torch.randn(T, B, 256) * 0.5
    with torch.no_grad():
        ref, _ = gru(x)  # (T, B, 2H)

    # kernel inputs
    w_ih = torch.cat([gru.weight_ih_l0, gru.weight_ih_l0_reverse], 0)  # (768,256)
    b_ih = torch.cat([gru.bias_ih_l0, gru.bias_ih_l0_reverse])
    xg = (x.reshape(T * B, 256) @ w_ih.t() + b_ih).view(T, B, 2, 384)
    u = torch.stack([gru.weight_hh_l0, gru.weight_hh_l0_reverse])
    bhh = torch.stack([gru.bias_hh_l0, gru.bias_hh_l0_reverse])

    ext = ops.ext()
    (hseq,) = ext.gru_layer_fwd(
        xg.to(torch.bfloat16).cuda().contiguous(),
        u.to(torch.bfloat16).cuda().contiguous(),
        bhh.float().cuda().contiguous(),
    )
    got = hseq.view(T, B, 256).float().cpu()
    err = (got - ref).abs()
    # bf16 recurrence over 90 steps: loose absolute tolerance, tight mean
    assert err.max().item() < 0.1, err.max().item()
    assert err.mean().item() < 0.01, err.mean().item()


@requires_gpu
def test_head_fwd_vs_torch():
    torch.manual_seed(3)
    T, B = 90, 8
    fc4 = torch.nn.Linear(256, 5)
    h = torch.randn(T, B, 256) * 0.3
    with torch.no_grad():
        ref = fc4(h).permute(1, 0, 2)  # (B, T, 5)
    ext = ops.ext()
    logits, amax = ext.head_fwd(
        h.to(torch.bfloat16).cuda().contiguous(),
        fc4.weight.detach().to(torch.bfloat16).cuda().contiguous(),
        fc4.bias.detach().float().cuda().contiguous(),
        True, True,
    )
    got = logits.cpu()
    assert (got - ref).abs().max().item() < 0.03
    agree = (amax.cpu().long() == ref.argmax(dim=2)).float().mean().item()
    assert agree > 0.995


@requires_gpu
def test_full_model_forward_matches_cpu():
    torch.manual_seed(4)
    m = RokoModel().eval()
    x = torch.randint(0, 12, (33, 200, 90))  # odd batch exercises padding
    with torch.no_grad():
        ref = m(x)  # CPU fp32 reference
        got = m.cuda()(x.cuda()).cpu()
    agree = (got.argmax(dim=2) == ref.argmax(dim=2)).float().mean().item()
    assert agree > 0.99, agree
    err = (got - ref).abs()
    assert err.mean().item() < 0.05, err.mean().item()


@requires_gpu
def test_gru_layer_bwd_vs_torch_autograd():
    """Custom GRU autograd (HIP fwd cache + BPTT kernel + GEMM reductions)
    vs torch.nn.GRU gradients on the same fp32 weights."""
    from roko_amd.ops.train import GruLayerFn

    torch.manual_seed(5)
    T, B = 90, 32
    gru = torch.nn.GRU(256, 128, num_layers=1, bidirectional=True)
    x = (torch.randn(T, B, 256) * 0.5).cuda().requires_grad_(True)

    gru_c = gru.cuda()
    ref_out, _ = gru_c(x)
    ref_loss = (ref_out * torch.linspace(0.5, 1.5, 256, device="cuda")).mean()
    ref_loss.backward()
    ref_gx = x.grad.clone()
    ref_gw = {n: p.grad.clone() for n, p in gru_c.named_parameters()}

    x.grad = None
    gru_c.zero_grad()
    w_ih = torch.cat([gru_c.weight_ih_l0, gru_c.weight_ih_l0_reverse], 0)
    b_ih = torch.cat([gru_c.bias_ih_l0, gru_c.bias_ih_l0_reverse])
    u = torch.stack([gru_c.weight_hh_l0, gru_c.weight_hh_l0_reverse])
    bhh = torch.stack([gru_c.bias_hh_l0, gru_c.bias_hh_l0_reverse])
    out = GruLayerFn.apply(x, w_ih, b_ih, u, bhh)
    loss = (out * torch.linspace(0.5, 1.5, 256, device="cuda")).mean()
    loss.backward()

    # bf16 kernels vs fp32 reference: compare directions via cosine + scale
    def close(a, b, name):
        a, b = a.reshape(-1).float(), b.reshape(-1).float()
        cos = torch.nn.functional.cosine_similarity(a, b, dim=0).item()
        rel = ((a - b).norm() / (b.norm() + 1e-12)).item()
        assert cos > 0.995, (name, cos)
        assert rel < 0.12, (name, rel)

    close(x.grad, ref_gx, "dx")
    close(gru_c.weight_ih_l0.grad, ref_gw["weight_ih_l0"], "dw_ih")
    close(gru_c.weight_hh_l0.grad, ref_gw["weight_hh_l0"], "dw_hh")
    close(gru_c.weight_hh_l0_reverse.grad, ref_gw["weight_hh_l0_reverse"], "dw_hh_r")
    close(gru_c.bias_hh_l0.grad, ref_gw["bias_hh_l0"], "dbhh")
    close(gru_c.bias_ih_l0.grad, ref_gw["bias_ih_l0"], "dbih")


@requires_gpu
def test_fused_ce_vs_torch():
    from roko_amd.ops.train import fused_cross_entropy

    torch.manual_seed(6)
    logits = (torch.randn(16, 90, 5, device="cuda") * 2).requires_grad_(True)
    y = torch.randint(0, 5, (16, 90), device="cuda")
    loss = fused_cross_entropy(logits, y)
    loss.backward()
    g1 = logits.grad.clone()

    logits2 = logits.detach().clone().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(logits2.transpose(1, 2), y)
    ref.backward()
    assert abs(loss.item() - ref.item()) < 1e-4
    assert torch.allclose(g1, logits2.grad, atol=1e-6)


@requires_gpu
def test_fused_adam_vs_torch():
    from roko_amd.ops.train import FusedAdam

    torch.manual_seed(7)
    p1 = torch.nn.Parameter(torch.randn(1000, device="cuda"))
    p2 = torch.nn.Parameter(torch.randn(333, device="cuda"))
    ref1 = torch.nn.Parameter(p1.detach().clone())
    ref2 = torch.nn.Parameter(p2.detach().clone())
    opt = FusedAdam([p1, p2], lr=1e-2)
    ref_opt = torch.optim.Adam([ref1, ref2], lr=1e-2)
    for it in range(5):
        g1 = torch.randn_like(p1)
        g2 = torch.randn_like(p2)
        opt.zero_grad()  # grad=None semantics: assign fresh tensors
        p1.grad = g1.clone()
        p2.grad = g2.clone()
        opt.step()
        ref_opt.zero_grad()
        ref1.grad = g1.clone()
        ref2.grad = g2.clone()
        ref_opt.step()
    assert torch.allclose(p1.detach(), ref1.detach(), atol=1e-5)
    assert torch.allclose(p2.detach(), ref2.detach(), atol=1e-5)


@requires_gpu
def test_full_train_step_loss_decreases():
    from roko_amd.ops.train import FusedAdam, fused_train_step

    torch.manual_seed(8)
    m = RokoModel().cuda().train()
    opt = FusedAdam(list(m.parameters()), lr=3e-3)
    x = torch.randint(0, 12, (32, 200, 90), dtype=torch.uint8, device="cuda")
    y = torch.randint(0, 5, (32, 90), device="cuda")
    losses = [float(fused_train_step(m, x, y, opt)) for _ in range(80)]
    # fixed random batch: the step must memorise it (dropout keeps it noisy)
    tail = sum(losses[-5:]) / 5
    head = sum(losses[:5]) / 5
    assert tail < head - 0.15, (head, tail)
    # and the generic train-mode model(x) also routes through the HIP GRU
    logits = m(x)
    assert logits.requires_grad
    assert logits.shape == (32, 90, 5)


@requires_gpu
def test_inference_pipeline_matches_argmax():
    """Pipelined hipGraph path must give the same predictions as the plain
    kernel path for every submitted batch (incl. a partial last batch)."""
    from roko_amd.ops.forward import InferencePipeline, roko_argmax

    torch.manual_seed(8)
    m = RokoModel().cuda().eval()
    pipe = InferencePipeline(m, batch=64, depth=3)
    xs = [torch.randint(0, 12, (64, 200, 90), dtype=torch.uint8, device="cuda")
          for _ in range(7)]
    xs.append(torch.randint(0, 12, (29, 200, 90), dtype=torch.uint8,
                            device="cuda"))  # partial batch
    tickets = [pipe.submit(x, copy_out=True) for x in xs]
    with torch.no_grad():
        refs = [roko_argmax(m, x).cpu() for x in xs]
    for t, ref in zip(tickets, refs):
        got = t()
        assert got.shape == ref.shape
        assert (got.long() == ref.long()).all()


@requires_gpu
def test_emb_grad_kernel_vs_scatter():
    """emb_grad reduction kernel vs autograd's embedding backward."""
    from roko_amd.ops.train import EmbedGatherFn

    torch.manual_seed(9)
    N = 7 * 200 * 90 + 13  # non-multiple of the kernel's row block
    ids = torch.randint(0, 12, (N,), dtype=torch.uint8, device="cuda")
    w = torch.randn(12, 50, device="cuda", requires_grad=True)
    out = EmbedGatherFn.apply(w, ids)
    dout = torch.randn_like(out.float()).to(torch.bfloat16)
    out.backward(dout)
    g_kernel = w.grad.clone()

    w2 = w.detach().clone().requires_grad_(True)
    ref = torch.nn.functional.embedding(ids.long(), w2.to(torch.bfloat16))
    ref.backward(dout)
    rel = (g_kernel - w2.grad).norm() / (w2.grad.norm() + 1e-9)
    assert rel.item() < 2e-2, rel.item()


@requires_gpu
def test_train_forward_matches_reference_grads():
    """Full differentiable GPU path (eval-mode dropout=identity) vs the
    fp32 autograd reference: logits and weight grads must agree."""
    from roko_amd.ops.train import train_forward

    torch.manual_seed(10)
    m = RokoModel().cuda().eval()  # dropout off -> comparable numerics
    x = torch.randint(0, 12, (32, 200, 90), device="cuda")
    y = torch.randint(0, 5, (32, 90), device="cuda")

    # MIOpen RNN cannot run backward in eval mode — use the native impl
    # for the reference graph
    with torch.backends.cudnn.flags(enabled=False):
        logits_ref = m._forward_torch(x.long())
        loss_ref = torch.nn.functional.cross_entropy(
            logits_ref.transpose(1, 2), y)
        m.zero_grad()
        loss_ref.backward()
    ref_g = {n: p.grad.clone() for n, p in m.named_parameters()}

    m.zero_grad()
    logits = train_forward(m, x)
    loss = torch.nn.functional.cross_entropy(logits.transpose(1, 2), y)
    loss.backward()

    assert abs(loss.item() - loss_ref.item()) < 5e-3
    for n, p in m.named_parameters():
        a, b = p.grad.reshape(-1).float(), ref_g[n].reshape(-1).float()
        cos = torch.nn.functional.cosine_similarity(a, b, dim=0).item()
        assert cos > 0.99, (n, cos)


@requires_gpu
def test_gemm_bias_vs_addmm():
    """Custom bf16 GEMM+bias kernel vs hipBLASLt for the GRU-projection
    shapes (incl. tails not divisible by the tile sizes)."""
    ext = ops.ext()
    torch.manual_seed(11)
    for M, N, K in [(11520, 768, 500), (11520, 500, 768), (2880, 768, 256),
                    (352, 130, 70)]:
        A = (torch.randn(M, K, device="cuda") * 0.3).to(torch.bfloat16)
        B = (torch.randn(K, N, device="cuda") * 0.3).to(torch.bfloat16)
        bias = torch.randn(N, device="cuda")
        got = ext.gemm_bias(A, B, bias).float()
        ref = torch.addmm(bias, A.float(), B.float())
        rel = (got - ref).norm() / ref.norm()
        assert rel.item() < 2e-2, (M, N, K, rel.item())
        got2 = ext.gemm_bias(A, B, None).float()
        ref2 = A.float() @ B.float()
        rel2 = (got2 - ref2).norm() / ref2.norm()
        assert rel2.item() < 2e-2, (M, N, K, rel2.item())


@requires_gpu
def test_gru_layer_fused_matches_split():
    """The serving fold (xg GEMM inside the GRU kernel) must match the
    split path (hipBLASLt addmm + gru_layer_fwd) for all three layer
    shapes within bf16 accumulation-order noise."""
    ext = ops.ext()
    torch.manual_seed(31)
    m = RokoModel().cuda().eval()
    w = fwd._bf16_weights(m)
    T, B = 90, 64
    x = (torch.randn(T, B, 500, device="cuda") * 0.4).to(torch.bfloat16)
    seq = x
    for l in range(3):
        xg = torch.addmm(
            w[f"b_ih{l}"], seq.reshape(T * B, -1), w[f"w_ih_t{l}"]
        ).view(T, B, 2, 384)
        (ref,) = ext.gru_layer_fwd(xg.contiguous(), w[f"u{l}"], w[f"bhh{l}"],
                                   False)
        got = ext.gru_layer_fused(seq.contiguous(), w[f"w_ih_p{l}"],
                                  w[f"b_ih{l}"], w[f"u{l}"], w[f"bhh{l}"])
        d = (got.float() - ref.float()).abs()
        assert d.max().item() < 0.03, (l, d.max().item())
        seq = ref.view(T, B, 256)


@requires_gpu
def test_atb_splitk_vs_mm():
    """Split-K A^T·B kernel vs hipBLASLt for the weight-grad shapes."""
    ext = ops.ext()
    torch.manual_seed(12)
    for K, M, N in [(11520, 384, 128), (11520, 768, 500), (2880, 768, 256),
                    (1000, 130, 70)]:
        A = (torch.randn(K, M, device="cuda") * 0.2).to(torch.bfloat16)
        B = (torch.randn(K, N, device="cuda") * 0.2).to(torch.bfloat16)
        got = ext.atb_splitk(A, B)
        ref = A.float().t() @ B.float()
        rel = (got - ref).norm() / ref.norm()
        assert rel.item() < 2e-2, (K, M, N, rel.item())


@requires_gpu
def test_graphed_train_step_matches_eager_and_learns():
    """Whole-iteration hipGraph capture: construction must leave the weights
    untouched (warmup state is snapshotted/restored — ADVICE r1), replays
    must track the eager fused step trajectory from the same init, advance
    the device counters once per replay, and learn a learnable task.

    Labels are the per-column consensus of the window (the real polishing
    signal) — purely random labels are NOT memorisable by this 1.27M-param
    model in 80 steps; the round-1 version of this test only passed because
    the warmup perturbation inflated losses[0]."""
    from roko_amd.ops.train import FusedAdam, GraphedTrainStep, fused_train_step

    torch.manual_seed(13)
    m1 = RokoModel().cuda().eval()  # dropout off: deterministic trajectories
    m2 = RokoModel().cuda().eval()
    m2.load_state_dict(m1.state_dict())
    o1 = FusedAdam(list(m1.parameters()), lr=2e-3)
    o2 = FusedAdam(list(m2.parameters()), lr=2e-3)
    p0 = o1.flat_p.clone()
    step = GraphedTrainStep(m1, o1, batch=32)
    assert torch.equal(o1.flat_p, p0), "capture warmup must not move weights"
    assert int(step.step_buf.item()) == 0
    assert torch.equal(o1.m, torch.zeros_like(o1.m))

    x = torch.randint(0, 12, (32, 200, 90), dtype=torch.uint8, device="cuda")
    y = torch.clamp(torch.mode(x.long() % 6, dim=1).values, max=4)  # consensus
    gl = [float(step(x, y)) for _ in range(60)]
    el = [float(fused_train_step(m2, x, y, o2)) for _ in range(60)]
    assert int(step.step_buf.item()) == 60
    # same math, two kernel schedules: trajectories must track closely
    for i in (0, 1, 5, 20, 59):
        assert abs(gl[i] - el[i]) < 5e-3 + 0.01 * el[i], (i, gl[i], el[i])
    # and the consensus task is learnable (uniform-random ids make the mode a
    # weak signal: expect a steady decrease, not memorisation — measured
    # 1.63 -> 1.44 over 60 steps on MI355X)
    assert gl[-1] < gl[0] - 0.08, (gl[0], gl[-1])
    assert el[-1] < el[0] - 0.08, (el[0], el[-1])


@requires_gpu
def test_dual_stream_step_matches_single():
    """Two half-batches on two streams must produce the same update as one
    full-batch step (eval-mode dropout off for exact comparison)."""
    from roko_amd.ops.train import (FusedAdam, dual_stream_train_step,
                                    fused_train_step)

    torch.manual_seed(14)
    m1 = RokoModel().cuda().eval()   # dropout off
    m2 = RokoModel().cuda().eval()
    m2.load_state_dict(m1.state_dict())
    o1 = FusedAdam(list(m1.parameters()), lr=1e-3)
    o2 = FusedAdam(list(m2.parameters()), lr=1e-3)
    x = torch.randint(0, 12, (64, 200, 90), dtype=torch.uint8, device="cuda")
    y = torch.randint(0, 5, (64, 90), device="cuda")

    l1 = fused_train_step(m1, x, y, o1)
    l2 = dual_stream_train_step(m2, x, y, o2)
    assert abs(float(l1) - float(l2)) < 2e-3, (float(l1), float(l2))
    for (n1, p1), (n2, p2) in zip(m1.named_parameters(), m2.named_parameters()):
        a, b = p1.detach().reshape(-1), p2.detach().reshape(-1)
        rel = ((a - b).norm() / (a.norm() + 1e-9)).item()
        assert rel < 5e-3, (n1, rel)




@requires_gpu
def test_deferred_weight_grads_match_inline():
    """The side-stream deferred weight-grad path (used by fused_train_step)
    produces the same gradients as the inline autograd path."""
    import numpy as np
    from roko_amd.model import RokoModel
    from roko_amd.ops.train import (deferred_weight_grads,
                                    drain_deferred_grads, fused_cross_entropy,
                                    train_forward)

    torch.manual_seed(0)
    model = RokoModel().cuda().train()
    g = torch.Generator().manual_seed(5)
    x = torch.randint(0, 12, (32, 200, 90), generator=g,
                      dtype=torch.uint8).cuda()
    y = torch.randint(0, 5, (32, 90), generator=g).cuda()

    def run(defer):
        model.zero_grad(set_to_none=True)
        torch.manual_seed(123)  # same dropout seeds both ways
        logits = train_forward(model, x)
        loss = fused_cross_entropy(logits, y)
        if defer:
            with deferred_weight_grads():
                loss.backward()
            drain_deferred_grads()
        else:
            loss.backward()
        torch.cuda.synchronize()
        return {n: p.grad.detach().clone()
                for n, p in model.named_parameters()}, float(loss)

    g0, l0 = run(False)
    g1, l1 = run(True)
    assert abs(l0 - l1) < 1e-5, (l0, l1)
    for n in g0:
        ref, got = g0[n].float(), g1[n].float()
        # 4e-3 relative: the two paths reduce bf16 inputs in different
        # orders (hipBLASLt GEMV vs split-K colsum), so 11520-term bias
        # sums differ at the bf16-ulp scale of the summands; real layout
        # bugs produce O(1) relative errors, far above this
        tol = 2e-4 + 4e-3 * ref.abs().max()
        assert (ref - got).abs().max() <= tol, (
            n, float((ref - got).abs().max()))


def test_xg_gemm2_numerics():
    """xg_gemm2 (the serving xg projection kernel) vs fp32 torch on both
    live shape families: l0 (K=500 padded to 512) and l1/l2 (K=256)."""
    from roko_amd.ops import _hip_ops as ext

    torch.manual_seed(0)
    M = 90 * 128
    for kreal, kp in [(500, 512), (256, 256)]:
        A = torch.randn(M, kreal, device="cuda").bfloat16().contiguous()
        W = (torch.randn(768, kreal, device="cuda") * 0.05).bfloat16()
        bias = (torch.randn(768, device="cuda") * 0.1).bfloat16()
        Wp = torch.zeros(768, kp, dtype=torch.bfloat16, device="cuda")
        Wp[:, :kreal] = W
        ref = torch.addmm(bias.float(), A.float(), W.t().float())
        out = ext.xg_gemm2(A, Wp.contiguous(), bias)
        rel = ((out.float() - ref).abs().max() / ref.abs().max()).item()
        assert rel < 2e-2, (kreal, rel)


def test_pipeline_parity_large_batch():
    """Large serving batches auto-enable the C++ hipGraph slot path
    (B >= 256); outputs must stay bit-exact vs the eager forward."""
    from roko_amd.ops.forward import InferencePipeline, roko_argmax

    model = RokoModel().cuda().eval()
    g = torch.Generator().manual_seed(11)
    for b in (256, 512):
        pipe = InferencePipeline(model, b, depth=3)
        x = torch.randint(0, 12, (b, 200, 90), generator=g,
                          dtype=torch.uint8).cuda()
        # several submits so the graph capture (2nd run) and replay paths
        # are both exercised on fresh inputs
        for _ in range(3):
            t = pipe.submit(x)
        got = t()
        ref = roko_argmax(model, x).cpu()
        assert torch.equal(got, ref), (b, (got != ref).float().mean())
        del pipe
