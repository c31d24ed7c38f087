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
    hseq = ext.gru_layer_fwd(
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
def test_model_refuses_silent_fallback():
    # training-mode GPU forward must not silently fall back to eager
    m = RokoModel().cuda().train()
    x = torch.randint(0, 12, (2, 200, 90)).cuda()
    with pytest.raises(RuntimeError):
        m(x)
