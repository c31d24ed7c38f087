"""GPU training path: fused front (FrontFn), custom-autograd GRU (HIP
fwd/BPTT kernels), fused cross-entropy, fused multi-tensor Adam.

Division of labour: the embedding/MLP front and the
sequential GRU recurrence run in hand-written HIP kernels; the weight-grad
reductions go to the split-K AtB kernel (dU, dW_ih) or hipBLASLt (xg, dx —
shapes where its tiles win). Dropout everywhere is counter-hash based so
the recompute backward regenerates masks without tensors.

Experimental steppers kept for the record: GraphedTrainStep (whole-step
hipGraph — measured parity with eager), dual_stream_train_step (half-batch
streams — host-launch-bound), GraphedDualTrainStep (their combination —
aborts in capture: the per-parameter AccumulateGrad node is SHARED by both
halves' graphs and its cross-stream gradient hand-off is capture-illegal).
The production step is fused_train_step.
"""

from __future__ import annotations

from typing import List, Optional

import torch

from .. import config as C


def _ext():
    from . import ext

    return ext()


_DEFER = False
_PENDING: list = []
_PENDING_FNS: list = []   # (ready_event, closure) run at drain time
_SIDE = None       # per-layer weight-grad GEMMs
_SIDE2 = None      # the front recompute backward (starts once the last GRU
                   # layer's dx exists; by then the main stream is idle)

# Cached scratch + output buffers for the raw weight-grad kernels
# (ext.gru_wgrads / ext.head_wgrads — one pybind call replaces the ~18-op
# aten section whose host enqueue cost ~1.3 ms/step, kernel trace tr5).
# Outputs ping-pong over TWO sets per key so a grad attached at drain time
# stays valid while the NEXT step's closure writes the other set (at most
# 2 deferred backwards may share a drain).
_WGRAD_BUFS: dict = {}


def _wgrad_bufs(key, dev, *shapes):
    """key -> {'pp': int, 'sets': [tuple(tensors), tuple(tensors)]}.
    shapes: (name, *dims) per output tensor; returns the next set."""
    ent = _WGRAD_BUFS.get(key)
    if ent is None:
        mk = lambda dims: torch.empty(*dims, device=dev, dtype=torch.float32)
        ent = {"pp": 0, "sets": [tuple(mk(d) for d in shapes),
                                 tuple(mk(d) for d in shapes)]}
        _WGRAD_BUFS[key] = ent
    ent["pp"] ^= 1
    return ent["sets"][ent["pp"]]


def _wgrad_ws(dev, slices, rows, cols):
    """Shared fp32 split-K workspace, grown on demand (scratch only —
    consumed within each gru_wgrads/head_wgrads call, safe to share
    across layers on one stream)."""
    key = ("ws", dev)
    ws = _WGRAD_BUFS.get(key)
    if (ws is None or ws.shape[0] < slices or ws.shape[1] < rows
            or ws.shape[2] < cols):
        ws = torch.empty(slices, rows, cols, device=dev,
                         dtype=torch.float32)
        _WGRAD_BUFS[key] = ws
    return ws


def _side_stream(dev) -> "torch.cuda.Stream":
    global _SIDE
    if _SIDE is None:
        # NOTE a hipExtStreamCreateWithCUMask stream (tried with 224 of 256
        # CUs for exactly this purpose) ran the same kernels ~3x slower and
        # cost the whole step 5.43 ms vs 3.67 ms — masked queues bypass the
        # normal scheduling path on this stack. Plain stream it is.
        _SIDE = torch.cuda.Stream(device=dev)
    return _SIDE


def _side2_stream(dev) -> "torch.cuda.Stream":
    global _SIDE2
    if _SIDE2 is None:
        _SIDE2 = torch.cuda.Stream(device=dev)
    return _SIDE2


class deferred_weight_grads:
    """Scope in which the weight-gradient reductions (split-K AtB kernels,
    bias GEMVs, the whole recompute front backward) are enqueued on a SIDE
    HIP stream instead of the autograd (main) stream, and parameters receive
    `.grad` at `drain_deferred_grads()`.

    Why: the BPTT/GRU kernels on the main stream are latency-bound at 8
    workgroups (3% of the chip), while the weight-grad GEMMs between them
    are wide — serial they add ~0.9 ms to the 3.85 ms step; on the side
    stream they fill the idle CUs under the recurrence (profiles/
    PERF_HISTORY.md). Opt-in (used by fused_train_step) so the default
    autograd semantics — p.grad populated by backward() — stay intact for
    every other caller."""

    def __enter__(self):
        global _DEFER
        # discard stashed grads from a step that died mid-backward — they
        # would otherwise be attached to the params on the next drain
        _PENDING.clear()
        _PENDING_FNS.clear()
        self._prev = _DEFER
        _DEFER = True
        return self

    def __exit__(self, *a):
        global _DEFER
        _DEFER = self._prev


def _defer_active() -> bool:
    return _DEFER and not torch.cuda.is_current_stream_capturing()


def _run_pending_fns():
    """Enqueue the stashed weight-grad closures on the side stream.

    The closures are HOST-deferred, not just stream-deferred: enqueueing
    the ~15 aten/kernel calls of one layer's weight grads costs ~250 us of
    host time, and doing it inside GruLayerFn.backward stalled the NEXT
    layer's BPTT launch (kernel-trace: 250-370 us main-stream gaps between
    gru_layer_bwd kernels). FrontFn.backward calls this right after
    enqueueing front_bwd, so both the host cost and the side-stream GPU
    work hide under the ~700 us front_bwd kernel; drain_deferred_grads
    keeps a fallback call for paths without a FrontFn."""
    if not _PENDING_FNS:
        return
    side = _side_stream(torch.cuda.current_device())
    for ev, fn in _PENDING_FNS:
        side.wait_event(ev)
        with torch.cuda.stream(side):
            fn()
    _PENDING_FNS.clear()


def drain_deferred_grads():
    """Run any remaining weight-grad closures, join the side streams, and
    attach the stashed gradients to params."""
    global _PENDING
    _run_pending_fns()
    if not _PENDING:
        return
    cur = torch.cuda.current_stream()
    if _SIDE is not None:
        cur.wait_stream(_SIDE)
    if _SIDE2 is not None:
        cur.wait_stream(_SIDE2)
    for p, g in _PENDING:
        g.record_stream(cur)
        if p.grad is None:
            p.grad = g
        else:
            p.grad = p.grad + g
    _PENDING = []


def train_step_available() -> bool:
    from . import available

    return available() and torch.cuda.is_available()


def fused_param_order(model) -> List[torch.nn.Parameter]:
    """Parameter order for FusedAdam that makes each GRU layer's forward /
    reverse tensors ADJACENT in the flat buffer: the per-step cat/stack
    packs in train_forward then become zero-copy views of flat_p (the
    three cat kernels measured ~80 us/step on the critical path)."""
    g = model.gru
    out = [model.embedding.weight, model.fc1.weight, model.fc1.bias,
           model.fc2.weight, model.fc2.bias, model.fc4.weight,
           model.fc4.bias]
    for l in range(C.NUM_LAYERS):
        for kind in ("weight_ih", "bias_ih", "weight_hh", "bias_hh"):
            out.append(getattr(g, f"{kind}_l{l}"))
            out.append(getattr(g, f"{kind}_l{l}_reverse"))
    return out


def _adjacent_view(a: torch.Tensor, b: torch.Tensor, shape):
    """A zero-copy tensor spanning two storage-adjacent tensors, or None."""
    if (a.untyped_storage().data_ptr() == b.untyped_storage().data_ptr()
            and a.storage_offset() + a.numel() == b.storage_offset()
            and a.is_contiguous() and b.is_contiguous()
            and a.dtype == b.dtype):
        t = torch.empty(0, dtype=a.dtype, device=a.device)
        t.set_(a.untyped_storage(), a.storage_offset(), shape)
        return t
    return None


def _packed_gru(model, l: int):
    """(w_ih (768, in), b_ih (768,), u (2, 384, 128), bhh (2, 384)) for
    layer l. Zero-copy flat_p views when the direction pairs are adjacent
    (FusedAdam over fused_param_order); fresh cats/stacks otherwise.
    Views are cached on the model (they track the live weights); cats are
    rebuilt every call (the weights change each step)."""
    cache = getattr(model, "_gru_pack_cache", None)
    if cache is None:
        cache = model._gru_pack_cache = {}
    hit = cache.get(l)
    g = model.gru
    wf = getattr(g, f"weight_ih_l{l}")
    if hit is not None and hit[0].untyped_storage().data_ptr() == \
            wf.untyped_storage().data_ptr():
        return hit
    wr = getattr(g, f"weight_ih_l{l}_reverse")
    bf_ = getattr(g, f"bias_ih_l{l}")
    br = getattr(g, f"bias_ih_l{l}_reverse")
    uf = getattr(g, f"weight_hh_l{l}")
    ur = getattr(g, f"weight_hh_l{l}_reverse")
    bhf = getattr(g, f"bias_hh_l{l}")
    bhr = getattr(g, f"bias_hh_l{l}_reverse")
    H = C.HIDDEN_SIZE
    w_ih = _adjacent_view(wf, wr, (6 * H, wf.shape[1]))
    b_ih = _adjacent_view(bf_, br, (6 * H,))
    u = _adjacent_view(uf, ur, (2, 3 * H, H))
    bhh = _adjacent_view(bhf, bhr, (2, 3 * H))
    if all(t is not None for t in (w_ih, b_ih, u, bhh)):
        # packed bf16 views from the FusedAdam flat mirror, when present
        # (same layout, same adjacency)
        mf = getattr(wf, "_rk_bf16", None)
        if mf is not None:
            w_ih._rk_bf16 = _adjacent_view(
                mf, wr._rk_bf16, (6 * H, wf.shape[1]))
            b_ih._rk_bf16 = _adjacent_view(bf_._rk_bf16, br._rk_bf16,
                                           (6 * H,))
            u._rk_bf16 = _adjacent_view(uf._rk_bf16, ur._rk_bf16,
                                        (2, 3 * H, H))
        cache[l] = (w_ih, b_ih, u, bhh)
        return cache[l]
    return (torch.cat([wf, wr], 0), torch.cat([bf_, br]),
            torch.stack([uf, ur]), torch.stack([bhf, bhr]))


def _bf(t: torch.Tensor) -> torch.Tensor:
    """bf16 image of a weight tensor: the FusedAdam flat-mirror view when
    present (refreshed once per step), else a fresh cast."""
    m = getattr(t, "_rk_bf16", None)
    return m if m is not None else t.detach().to(torch.bfloat16)


def _accum_pref_grads(prefs, dw_ih, db_ih, du, dbhh) -> None:
    """Route the GRU weight grads straight onto the underlying parameters
    (the packed w_ih/u inputs may be flat_p views with no autograd
    lineage). In-place add when .grad exists (keeps GraphedTrainStep's
    flat-grad view addresses stable across replays)."""
    wf, wr, bf_, br, uf, ur, bhf, bhr = prefs
    G3 = 3 * C.HIDDEN_SIZE
    for p, g_ in ((wf, dw_ih[:G3]), (wr, dw_ih[G3:]),
                  (bf_, db_ih[:G3]), (br, db_ih[G3:]),
                  (uf, du[0]), (ur, du[1]), (bhf, dbhh[0]), (bhr, dbhh[1])):
        if p.grad is None:
            p.grad = g_.contiguous()
        else:
            p.grad.add_(g_)


class GruLayerFn(torch.autograd.Function):
    """One bidirectional GRU layer over (T, B, in) through the HIP kernels."""

    @staticmethod
    def forward(ctx, x_seq, w_ih, b_ih_all, u, bhh, prefs=None):
        # x_seq (T, B, in) fp32/bf16; w_ih (768, in); b_ih_all (768,)
        # u (2, 384, 128); bhh (2, 384); prefs: the 8 underlying per-direction
        # parameters (w_ih f/r, b_ih f/r, w_hh f/r, b_hh f/r) for the
        # deferred-weight-grad path (w_ih/u here are cat/stack composites)
        ext = _ext()
        T, B, _ = x_seq.shape
        x_bf = x_seq.to(torch.bfloat16)
        w_ih_bf = _bf(w_ih)
        xg = torch.addmm(
            _bf(b_ih_all),
            x_bf.reshape(T * B, -1),
            w_ih_bf.t(),
        ).view(T, B, 2, 384).contiguous()
        u_bf = _bf(u).contiguous()
        bhh_f = bhh.detach().float().contiguous()
        hseq, cache = ext.gru_layer_fwd(xg, u_bf, bhh_f, True)
        ctx.save_for_backward(x_bf, w_ih_bf, u_bf, hseq, cache)
        ctx.in_dtype = x_seq.dtype
        ctx.prefs = prefs
        return hseq.view(T, B, 2 * C.HIDDEN_SIZE)

    @staticmethod
    def backward(ctx, dout):
        ext = _ext()
        x_bf, w_ih_bf, u_bf, hseq, cache = ctx.saved_tensors
        T, B, _ = x_bf.shape
        H = C.HIDDEN_SIZE
        dhin = dout.reshape(T, B, 2, H).to(torch.bfloat16).contiguous()
        ut = u_bf.transpose(1, 2).contiguous()  # (2, 128, 384)
        # kernel emits GEMM-ready layouts: dxg (T,B,2,384) -> (TB,768) view;
        # dhg (2,T,B,384) -> contiguous per-direction (TB,384) slices
        dxg, dhg = ext.gru_layer_bwd(cache, hseq, dhin, ut)

        TB = T * B
        dxg_cat = dxg.reshape(TB, 2 * 3 * H)  # zero-copy (TB, 768)
        dx = dxg_cat.mm(w_ih_bf).to(ctx.in_dtype).view(T, B, -1)

        def weight_grads():
            # h_prev sequences per direction
            zeros = hseq.new_zeros(1, B, H)
            hp_f = torch.cat([zeros, hseq[:-1, :, 0, :]], dim=0)  # (T,B,H)
            hp_r = torch.cat([hseq[1:, :, 1, :], zeros], dim=0)
            dhg_f = dhg[0].reshape(TB, 3 * H)
            dhg_r = dhg[1].reshape(TB, 3 * H)
            # custom split-K A^T·B kernels: hipBLASLt schedules these
            # K=11520 transpose-A reductions on 6 workgroups (81 us each)
            du = torch.stack(
                [ext.atb_splitk(dhg_f, hp_f.reshape(TB, H).contiguous()),
                 ext.atb_splitk(dhg_r, hp_r.reshape(TB, H).contiguous())]
            )
            x_flat = x_bf.reshape(TB, -1)
            dw_ih = ext.atb_splitk(dxg_cat, x_flat.contiguous())  # (768,in)
            # column sums as one hipBLASLt GEMV each
            ones = x_bf.new_ones(1, TB)
            dbhh = torch.stack(
                [ones.mm(dhg_f).squeeze(0), ones.mm(dhg_r).squeeze(0)]
            ).float()
            db_ih = ones.mm(dxg_cat).squeeze(0).float()  # (768,)
            return du, dw_ih, dbhh, db_ih

        if _defer_active() and ctx.prefs is not None:
            # host-defer the whole weight-grad section: enqueueing it here
            # stalls the next layer's BPTT launch. The closure runs on the
            # side stream at drain_deferred_grads(), ordered after this
            # point by the event — and it is ONE pybind call into raw
            # kernels (ext.gru_wgrads): the aten form (cats, contiguous
            # copies, ones-GEMVs) cost ~1.3 ms/step of host enqueue.
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream())
            prefs = ctx.prefs
            G3 = 3 * H
            IN = x_bf.shape[-1]

            def deferred():
                du, dw_ih, dbhh, db_ih = _wgrad_bufs(
                    id(prefs[0]), dx.device,
                    (2, G3, H), (2 * G3, IN), (2, G3), (2 * G3,))
                ws = _wgrad_ws(dx.device, (TB + 255) // 256, 2 * G3,
                               max(IN, 128))
                ext.gru_wgrads(dhg, dxg_cat, hseq, x_bf.reshape(TB, IN),
                               ws, ws, du, dw_ih, dbhh, db_ih)
                side = torch.cuda.current_stream()
                for t in (hseq, dhg, dxg, x_bf):
                    t.record_stream(side)
                wf, wr, bf_, br, uf, ur, bhf, bhr = prefs
                _PENDING.extend([
                    (wf, dw_ih[:G3]), (wr, dw_ih[G3:]),
                    (bf_, db_ih[:G3]), (br, db_ih[G3:]),
                    (uf, du[0]), (ur, du[1]),
                    (bhf, dbhh[0]), (bhr, dbhh[1]),
                ])

            _PENDING_FNS.append((ev, deferred))
            return dx, None, None, None, None, None

        du, dw_ih, dbhh, db_ih = weight_grads()
        if ctx.prefs is not None:
            # the packed inputs may be lineage-free flat_p views: put the
            # weight grads straight on the real parameters
            _accum_pref_grads(ctx.prefs, dw_ih, db_ih, du, dbhh)
            return dx, None, None, None, None, None
        return dx, dw_ih, db_ih, du, dbhh, None


class HeadFn(torch.autograd.Function):
    """fc4 head through the fused GEMV kernel (fp32 logits straight from
    bf16 activations) with a bf16 backward: replaces the fp32 aten linear
    chain (~150 us/step of skinny fp32 GEMMs plus an 11.8 MB (T,B,256)
    fp32 activation copy — profiles/train_r02_final_kernel_stats.csv).
    Reference op: rnn_model.py:59."""

    @staticmethod
    def forward(ctx, seq, w4, b4):
        ext = _ext()
        w4b = _bf(w4).contiguous()
        (logits,) = ext.head_fwd(seq.contiguous(), w4b,
                                 b4.detach().float().contiguous(),
                                 True, False)
        ctx.save_for_backward(seq, w4b)
        ctx.prefs = (w4, b4)
        return logits  # (B, T, 5) fp32

    @staticmethod
    def backward(ctx, dlogits):
        ext = _ext()
        seq, w4b = ctx.saved_tensors
        T, B, _ = seq.shape
        dl = (dlogits.permute(1, 0, 2).reshape(T * B, C.NUM_CLASSES)
              .to(torch.bfloat16).contiguous())
        dseq = dl.mm(w4b).view(T, B, 2 * C.HIDDEN_SIZE)

        def wgrads():
            dw4 = ext.atb_splitk(dl,
                                 seq.reshape(T * B, 2 * C.HIDDEN_SIZE)
                                 .contiguous())
            ones = dl.new_ones(1, T * B)
            db4 = ones.mm(dl).squeeze(0).float()
            return dw4, db4

        w4, b4 = ctx.prefs
        if _defer_active():
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream())

            def deferred():
                dw4, db4 = _wgrad_bufs(id(w4), dseq.device,
                                       (C.NUM_CLASSES, 2 * C.HIDDEN_SIZE),
                                       (C.NUM_CLASSES,))
                ws = _wgrad_ws(dseq.device, (T * B + 255) // 256,
                               2 * 3 * C.HIDDEN_SIZE,
                               max(2 * C.HIDDEN_SIZE, 128))
                ext.head_wgrads(dl, seq.reshape(T * B, 2 * C.HIDDEN_SIZE),
                                ws, dw4, db4)
                side = torch.cuda.current_stream()
                for t in (dl, seq):
                    t.record_stream(side)
                _PENDING.extend([(w4, dw4), (b4, db4)])

            _PENDING_FNS.append((ev, deferred))
            return dseq, None, None
        dw4, db4 = wgrads()
        for p, g_ in ((w4, dw4), (b4, db4)):
            if p.grad is None:
                p.grad = g_.contiguous()
            else:
                p.grad.add_(g_)
        return dseq, None, None


class FusedCrossEntropy(torch.autograd.Function):
    """Mean CE over (N, 5) logits: fwd loss and dlogits in one kernel pass."""

    @staticmethod
    def forward(ctx, logits, target):
        ext = _ext()
        loss, dlogits = ext.ce_fwd_bwd(
            logits.reshape(-1, C.NUM_CLASSES).float().contiguous(),
            target.reshape(-1).contiguous(),
        )
        ctx.save_for_backward(dlogits)
        ctx.shape = logits.shape
        return loss

    @staticmethod
    def backward(ctx, grad_out):
        (dlogits,) = ctx.saved_tensors
        return (grad_out * dlogits).view(ctx.shape), None


def fused_cross_entropy(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """logits (B, W, 5), target (B, W) int64 -> scalar mean loss."""
    return FusedCrossEntropy.apply(logits, target)


class FusedAdam:
    """Flat-buffer Adam with grad=None accumulation semantics: parameters and
    moments live in flat fp32 tensors (param tensors are views); gradients
    stay per-tensor (autograd ASSIGNS them — no AccumulateGrad add kernel per
    parameter, which cost ~123 us/step as 26 tiny launches) and the update is
    one multi-tensor kernel driven by a [grad_ptr, offset, numel] table. In
    DP a single gather kernel packs the grads into one flat tensor for ONE
    all-reduce.

    On CPU (no HIP extension) the same flat-buffer semantics run through
    plain torch ops, so the DP sync path (`allreduce_grads`) is exercised by
    the multi-process gloo tests byte-for-byte as the GPU path would be."""

    def __init__(self, params: List[torch.nn.Parameter], lr: float = C.LR,
                 betas=(0.9, 0.999), eps: float = 1e-8):
        self.params = [p for p in params if p.requires_grad]
        self.lr, self.betas, self.eps = lr, betas, eps
        self.step_count = 0
        n = sum(p.numel() for p in self.params)
        dev = self.params[0].device
        self.on_gpu = dev.type == "cuda"
        self.flat_p = torch.empty(n, dtype=torch.float32, device=dev)
        self.flat_g = torch.zeros(n, dtype=torch.float32, device=dev)
        self.m = torch.zeros(n, dtype=torch.float32, device=dev)
        self.v = torch.zeros(n, dtype=torch.float32, device=dev)
        self.offs = []
        off = 0
        for p in self.params:
            k = p.numel()
            self.flat_p[off : off + k].copy_(p.data.reshape(-1))
            p.data = self.flat_p[off : off + k].view(p.shape)
            self.offs.append(off)
            off += k

    def zero_grad(self):
        for p in self.params:
            p.grad = None

    def refresh_bf16(self):
        """One flat fp32->bf16 cast of every parameter; per-param bf16
        views hang off the params as ``_rk_bf16`` (train_forward uses them
        instead of ~26 small per-tensor .to(bfloat16) kernels per step).
        Call after step() / before the next forward."""
        if not hasattr(self, "flat_bf"):
            self.flat_bf = torch.empty_like(self.flat_p,
                                            dtype=torch.bfloat16)
            for p, off in zip(self.params, self.offs):
                p._rk_bf16 = self.flat_bf[off : off + p.numel()].view(p.shape)
        self.flat_bf.copy_(self.flat_p)

    def state_dict(self) -> dict:
        """Resume sidecar payload (roko_amd.train CheckpointManager)."""
        return {
            "kind": "fused_adam",
            "step_count": self.step_count,
            "m": self.m.detach().cpu(),
            "v": self.v.detach().cpu(),
            "lr": self.lr, "betas": self.betas, "eps": self.eps,
        }

    def load_state_dict(self, st: dict) -> None:
        if st.get("kind") != "fused_adam":
            raise ValueError("not a FusedAdam state dict")
        self.step_count = int(st["step_count"])
        self.m.copy_(st["m"].to(self.m.device))
        self.v.copy_(st["v"].to(self.v.device))
        self.lr = st.get("lr", self.lr)
        self.betas = tuple(st.get("betas", self.betas))
        self.eps = st.get("eps", self.eps)

    def _gather_flat(self) -> None:
        """Pack per-param grads into flat_g (one kernel on GPU)."""
        if self.on_gpu:
            tab = self._table()
            _ext().grad_gather(tab, len(self.params), self.flat_g)
            return
        for p, off in zip(self.params, self.offs):
            assert p.grad is not None, "param missing grad"
            self.flat_g[off : off + p.numel()].copy_(
                p.grad.detach().reshape(-1).float())

    def _torch_adam(self, g: torch.Tensor) -> None:
        """CPU fallback of the fused update, same math as adam.hip."""
        b1, b2 = self.betas
        self.m.mul_(b1).add_(g, alpha=1 - b1)
        self.v.mul_(b2).addcmul_(g, g, value=1 - b2)
        bc1 = 1 - b1 ** self.step_count
        bc2 = 1 - b2 ** self.step_count
        denom = (self.v / bc2).sqrt_().add_(self.eps)
        self.flat_p.addcdiv_(self.m / bc1, denom, value=-self.lr)

    def _flat_g2(self):
        if not hasattr(self, "_g2"):
            self._g2 = torch.empty_like(self.flat_g)
        return self._g2

    def bind_flat_grads(self):
        """Captured mode: gradients become views of the flat buffer so the
        graph records accumulation into STABLE addresses; zeroing is one
        fill on flat_g."""
        for p, off in zip(self.params, self.offs):
            p.grad = self.flat_g[off : off + p.numel()].view(p.shape)

    def _table(self):
        rows = []
        for p, off in zip(self.params, self.offs):
            g = p.grad
            assert g is not None, "param missing grad"
            if g.dtype != torch.float32 or not g.is_contiguous():
                g = g.float().contiguous()
                p.grad = g
            rows.append((g.data_ptr(), off, g.numel()))
        cpu = torch.tensor(rows, dtype=torch.int64)
        return cpu.to(self.flat_p.device, non_blocking=True)

    def allreduce_grads(self):
        import torch.distributed as dist

        if dist.is_initialized() and dist.get_world_size() > 1:
            self._gather_flat()
            dist.all_reduce(self.flat_g)
            self.flat_g /= dist.get_world_size()
            self._synced = True
        else:
            self._synced = False

    def step(self):
        self.step_count += 1
        if not self.on_gpu:
            if not getattr(self, "_synced", False):
                self._gather_flat()
            self._torch_adam(self.flat_g)
        elif getattr(self, "_synced", False):
            _ext().adam_step(self.flat_p, self.flat_g, self.m, self.v, self.lr,
                             self.betas[0], self.betas[1], self.eps,
                             self.step_count)
        else:
            tab = self._table()
            _ext().adam_mt(tab, len(self.params), self.flat_p, self.m, self.v,
                           self.lr, self.betas[0], self.betas[1], self.eps,
                           self.step_count)


class FrontFn(torch.autograd.Function):
    """Fused embedding+MLP training front (fwd + recompute backward in two
    HIP kernels — front_train.hip). Dropout masks are counter-based hashes
    of (seed, index); the backward regenerates them, so no mask tensors and
    no activation round-trips through HBM."""

    @staticmethod
    def forward(ctx, emb_w, w1, b1, w2, b2, ids_u8, p_drop, training,
                seed_buf=None):
        ext = _ext()
        keep = 1.0 - (p_drop if training else 0.0)
        # hipGraph capture: the seed lives in a device buffer advanced by a
        # captured add, so every replay gets fresh dropout masks with no
        # host RNG (a host randint would freeze the mask into the graph)
        seed = 0 if seed_buf is not None else int(
            torch.randint(0, 2**31 - 1, (1,)).item())
        w1b = _bf(w1).contiguous()
        b1f = b1.detach().float().contiguous()
        w2b = _bf(w2).contiguous()
        b2f = b2.detach().float().contiguous()
        embb = _bf(emb_w).contiguous()
        out = ext.front_fwd(ids_u8, w1b, b1f, w2b, b2f, embb, seed, keep,
                            seed_buf)
        ctx.save_for_backward(ids_u8, w1b, b1f, w2b, b2f, embb)
        ctx.seed, ctx.keep, ctx.seed_buf = seed, keep, seed_buf
        ctx.prefs = (emb_w, w1, b1, w2, b2)
        return out  # (90, B, 500) bf16

    @staticmethod
    def backward(ctx, dseq):
        ext = _ext()
        ids_u8, w1b, b1f, w2b, b2f, embb = ctx.saved_tensors
        if _defer_active():
            # the front produces ONLY weight gradients (it is the first
            # layer): the whole recompute backward runs on the side stream,
            # overlapping the GRU BPTT kernels still on the main stream.
            # Enqueue the stashed GRU/head weight-grad closures FIRST:
            # front_bwd fills the whole chip, so side-stream kernels cannot
            # co-run with it (kernel trace: a 12 us slice_sum stretched to
            # 685 us under it) — their only overlap window is the 8-WG BPTT
            # region still draining on the main stream.
            _run_pending_fns()
            cur = torch.cuda.current_stream()
            side = _side2_stream(dseq.device)
            side.wait_stream(cur)
            with torch.cuda.stream(side):
                de, dw1, db1, dw2, db2 = ext.front_bwd(
                    ids_u8, dseq.to(torch.bfloat16).contiguous(), w1b, b1f,
                    w2b, b2f, embb, ctx.seed, ctx.keep, 0x1F, ctx.seed_buf,
                )
            for t in (ids_u8, w1b, b1f, w2b, b2f, embb, dseq):
                t.record_stream(side)
            _PENDING.extend(zip(ctx.prefs, (de, dw1, db1, dw2, db2)))
            return None, None, None, None, None, None, None, None, None
        de, dw1, db1, dw2, db2 = ext.front_bwd(
            ids_u8, dseq.to(torch.bfloat16).contiguous(), w1b, b1f, w2b, b2f,
            embb, ctx.seed, ctx.keep, 0x1F, ctx.seed_buf,
        )
        return de, dw1, db1, dw2, db2, None, None, None, None


class EmbedGatherFn(torch.autograd.Function):
    """Embedding gather whose backward is the fused (12, 50) LDS reduction
    kernel instead of aten's 115M-element scatter-add (9.1 ms/step) or a
    skinny-K Tensile GEMM (1.65 ms/step) — profiles/train_breakdown_r01."""

    @staticmethod
    def forward(ctx, weight, ids_u8):
        ctx.save_for_backward(ids_u8)
        return torch.nn.functional.embedding(
            ids_u8.int(), weight.to(torch.bfloat16)
        )

    @staticmethod
    def backward(ctx, dout):
        (ids_u8,) = ctx.saved_tensors
        de = _ext().emb_grad(
            dout.reshape(-1, C.EMBED_DIM).contiguous(), ids_u8.reshape(-1)
        )
        return de, None


def train_forward(model, x: torch.Tensor, seed_buf=None) -> torch.Tensor:
    """Differentiable training forward on GPU: bf16 torch GEMMs for the MLP
    front (keeps the reference's dropout semantics), HIP kernels for the
    embedding backward and the GRU.

    The whole front (embedding gather, three dropouts, fc1/fc2 GEMMs, relu,
    and the permute into GRU layout) is TWO kernel launches via FrontFn —
    the reference's semantics (rnn_model.py:47-56) with hash-based dropout
    masks instead of materialised ones."""
    ids = x.to(torch.uint8)
    seq = FrontFn.apply(
        model.embedding.weight, model.fc1.weight, model.fc1.bias,
        model.fc2.weight, model.fc2.bias, ids, float(model.dropout.p),
        model.training, seed_buf,
    )
    # (T, B, 500) bf16
    g = model.gru
    drop_p = g.dropout if model.training else 0.0
    for l in range(C.NUM_LAYERS):
        w_ih, b_ih, u, bhh = _packed_gru(model, l)
        if l > 0 and drop_p > 0:
            seq = torch.nn.functional.dropout(seq, drop_p, model.training)
        prefs = (
            getattr(g, f"weight_ih_l{l}"), getattr(g, f"weight_ih_l{l}_reverse"),
            getattr(g, f"bias_ih_l{l}"), getattr(g, f"bias_ih_l{l}_reverse"),
            getattr(g, f"weight_hh_l{l}"), getattr(g, f"weight_hh_l{l}_reverse"),
            getattr(g, f"bias_hh_l{l}"), getattr(g, f"bias_hh_l{l}_reverse"),
        )
        seq = GruLayerFn.apply(seq, w_ih, b_ih, u, bhh, prefs)
    logits = HeadFn.apply(seq, model.fc4.weight, model.fc4.bias)  # (B, T, 5)
    return logits


def fused_train_step(model, x, y, opt: Optional[FusedAdam] = None,
                     reducer=None) -> torch.Tensor:
    """One full training step through the fused path; returns the loss.

    DP gradient sync happens inside ``opt.allreduce_grads()`` (FusedAdam) —
    a GradReducer's autograd hooks never fire here because the fused
    backward ASSIGNS grads, so passing one would silently skip the sync."""
    if reducer is not None and getattr(reducer, "enabled", False):
        raise ValueError(
            "fused_train_step does not support GradReducer; DP sync runs "
            "through FusedAdam.allreduce_grads (pass opt=FusedAdam(...))")
    if opt is not None and opt.on_gpu:
        opt.refresh_bf16()
    logits = train_forward(model, x)
    loss = fused_cross_entropy(logits, y)
    if opt is not None:
        opt.zero_grad()
    else:
        model.zero_grad(set_to_none=False)
    with deferred_weight_grads():
        loss.backward()
    drain_deferred_grads()
    if opt is not None:
        opt.allreduce_grads()
        opt.step()
    return loss.detach()


class GraphedTrainStep:
    """Whole-iteration hipGraph capture: dropout-seed bump, forward, fused
    cross-entropy, backward, (DP all-reduce,) fused Adam — one graph replay
    per step instead of ~60 kernel launches. Gradients are flat-buffer views
    (stable addresses across replays); the dropout seed and Adam step live
    in device buffers advanced by captured adds.

    Falls back cleanly: construct inside try/except and use
    fused_train_step when capture raises.
    """

    def __init__(self, model, opt: FusedAdam, batch: int, world: int = 1):
        import torch.distributed as dist

        dev = next(model.parameters()).device
        self.x = torch.zeros((batch, C.WINDOW_ROWS, C.WINDOW_COLS),
                             dtype=torch.uint8, device=dev)
        self.y = torch.zeros((batch, C.WINDOW_COLS), dtype=torch.int64,
                             device=dev)
        self.seed_buf = torch.randint(0, 2**31 - 1, (1,), dtype=torch.int32,
                                      device=dev)
        self.step_buf = torch.zeros(1, dtype=torch.int32, device=dev)
        self.opt = opt
        self.world = world
        opt.bind_flat_grads()

        def one_step():
            self.seed_buf.add_(747796405)  # odd constant: full-period walk
            self.step_buf.add_(1)
            opt.flat_g.zero_()
            logits = train_forward(model, self.x, seed_buf=self.seed_buf)
            loss = fused_cross_entropy(logits, self.y)
            loss.backward()
            if world > 1:
                dist.all_reduce(opt.flat_g)
                opt.flat_g /= world
            _ext().adam_step(opt.flat_p, opt.flat_g, opt.m, opt.v, opt.lr,
                             opt.betas[0], opt.betas[1], opt.eps, 0,
                             self.step_buf)
            return loss

        # warmup runs one_step() for real — snapshot optimizer/model state so
        # training starts from the caller's weights and step 0, not from 3
        # all-zero-batch Adam updates (restored after capture; capture itself
        # only records, it does not execute)
        snap = (opt.flat_p.clone(), opt.m.clone(), opt.v.clone(),
                self.seed_buf.clone())
        # warmup on a side stream (torch full-network capture recipe)
        s = torch.cuda.Stream(device=dev)
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                one_step()
        torch.cuda.current_stream().wait_stream(s)
        # a GC run during capture can free non-pool GPU memory (dead graphs,
        # pools from earlier work) -> hipFree inside capture -> abort in a
        # destructor. Reclaim garbage now and hold GC for the capture.
        import gc

        torch.cuda.synchronize()
        gc.collect()
        gc_was_enabled = gc.isenabled()
        gc.disable()
        try:
            self.graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.graph):
                self.loss = one_step()
        finally:
            if gc_was_enabled:
                gc.enable()
        # undo the warmup's 3 real updates (see snapshot above)
        opt.flat_p.copy_(snap[0])
        opt.m.copy_(snap[1])
        opt.v.copy_(snap[2])
        self.seed_buf.copy_(snap[3])
        self.step_buf.zero_()
        torch.cuda.synchronize()

    def __call__(self, x, y):
        self.x.copy_(x, non_blocking=True)
        self.y.copy_(y, non_blocking=True)
        self.graph.replay()
        return self.loss


def dual_stream_train_step(model, x, y, opt: FusedAdam, streams=None):
    """One optimizer step over batch (x, y) as TWO half-batches on two HIP
    streams. The GRU recurrence is latency-bound at ~3% chip occupancy
    (8 workgroups), so one half's sequential GRU chain overlaps the other
    half's front/GEMM work almost for free; halving the batch does not
    change the GRU kernels' duration (fewer workgroups, same 90 steps).

    Race-free gradient handling without replicas: after the first
    backward's launches are enqueued, its grad TENSORS are harvested and
    p.grad reset to None, so the second backward ASSIGNS fresh tensors —
    the two halves' grads live in different buffers and are merged with two
    gather kernels into the flat buffer ((gA + gB) / 2: each half-loss is a
    mean over its half).
    """
    import torch.distributed as dist

    dev = x.device
    if streams is None:
        streams = (torch.cuda.Stream(device=dev), torch.cuda.Stream(device=dev))
    s0, s1 = streams
    B = x.shape[0]
    h = B // 2
    cur = torch.cuda.current_stream()
    s0.wait_stream(cur)
    s1.wait_stream(cur)

    opt.zero_grad()
    opt.refresh_bf16()
    with torch.cuda.stream(s0):
        logits0 = train_forward(model, x[:h])
        loss0 = fused_cross_entropy(logits0, y[:h])
        loss0.backward()
    grads_a = []
    for p in opt.params:
        grads_a.append(p.grad)
        p.grad = None
    with torch.cuda.stream(s1):
        logits1 = train_forward(model, x[h:])
        loss1 = fused_cross_entropy(logits1, y[h:])
        loss1.backward()

    cur.wait_stream(s0)
    cur.wait_stream(s1)
    # merge: flat_g = (gA + gB) / 2, then (all-reduce and) Adam
    ext = _ext()

    def table_of(grads):
        rows = []
        for g, off in zip(grads, opt.offs):
            assert g is not None
            if g.dtype != torch.float32 or not g.is_contiguous():
                g = g.float().contiguous()
            rows.append((g.data_ptr(), off, g.numel()))
        return torch.tensor(rows, dtype=torch.int64).to(dev, non_blocking=True)

    tab_a = table_of(grads_a)
    grads_b = [p.grad for p in opt.params]
    tab_b = table_of(grads_b)
    for g in grads_a:
        g.record_stream(cur)
    for g in grads_b:
        g.record_stream(cur)
    ext.grad_gather(tab_a, len(opt.params), opt.flat_g)
    # second gather adds? grad_gather overwrites — accumulate via temp
    ext.grad_gather(tab_b, len(opt.params), opt._flat_g2())
    opt.flat_g.add_(opt._flat_g2()).mul_(0.5)
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.all_reduce(opt.flat_g)
        opt.flat_g /= dist.get_world_size()
    opt.step_count += 1
    ext.adam_step(opt.flat_p, opt.flat_g, opt.m, opt.v, opt.lr, opt.betas[0],
                  opt.betas[1], opt.eps, opt.step_count)
    # keep the harvested tensors alive until the merge kernels ran
    return (loss0.detach() + loss1.detach()) * 0.5


class GraphedDualTrainStep:
    """EXPERIMENTAL — currently aborts during capture: both halves share
    each parameter's AccumulateGrad node, whose cross-stream gradient
    hand-off is illegal inside stream capture (torch warns, HIP aborts).
    Kept as the documented starting point for a replica-leaf variant.

    Design: dual-stream half-batch step captured as ONE hipGraph — the
    eager dual-stream variant wins GRU-latency overlap but loses to doubled
    host launch cost; capture would remove the host from the loop."""

    def __init__(self, model, opt: FusedAdam, batch: int, world: int = 1):
        import torch.distributed as dist

        dev = next(model.parameters()).device
        h = batch // 2
        self.x = torch.zeros((batch, C.WINDOW_ROWS, C.WINDOW_COLS),
                             dtype=torch.uint8, device=dev)
        self.y = torch.zeros((batch, C.WINDOW_COLS), dtype=torch.int64,
                             device=dev)
        self.seed0 = torch.randint(0, 2**31 - 1, (1,), dtype=torch.int32,
                                   device=dev)
        self.seed1 = torch.randint(0, 2**31 - 1, (1,), dtype=torch.int32,
                                   device=dev)
        self.step_buf = torch.zeros(1, dtype=torch.int32, device=dev)
        P = len(opt.params)
        self.tab_a = torch.zeros((P, 3), dtype=torch.int64, device=dev)
        self.tab_b = torch.zeros((P, 3), dtype=torch.int64, device=dev)
        self.opt = opt
        s0 = torch.cuda.Stream(device=dev)
        s1 = torch.cuda.Stream(device=dev)
        ext = _ext()

        def host_table(grads):
            rows = [(g.data_ptr(), off, g.numel())
                    for g, off in zip(grads, opt.offs)]
            return torch.tensor(rows, dtype=torch.int64).to(dev)

        def one_step(static_tabs: bool):
            self.seed0.add_(747796405)
            self.seed1.add_(931541387)
            self.step_buf.add_(1)
            cur = torch.cuda.current_stream()
            s0.wait_stream(cur)
            s1.wait_stream(cur)
            opt.zero_grad()
            with torch.cuda.stream(s0):
                logits0 = train_forward(model, self.x[:h],
                                        seed_buf=self.seed0)
                loss0 = fused_cross_entropy(logits0, self.y[:h])
                loss0.backward()
            grads_a = [p.grad for p in opt.params]
            for p in opt.params:
                p.grad = None
            with torch.cuda.stream(s1):
                logits1 = train_forward(model, self.x[h:],
                                        seed_buf=self.seed1)
                loss1 = fused_cross_entropy(logits1, self.y[h:])
                loss1.backward()
            grads_b = [p.grad for p in opt.params]
            cur.wait_stream(s0)
            cur.wait_stream(s1)
            ta = self.tab_a if static_tabs else host_table(grads_a)
            tb = self.tab_b if static_tabs else host_table(grads_b)
            ext.grad_gather(ta, len(opt.params), opt.flat_g)
            ext.grad_gather(tb, len(opt.params), opt._flat_g2())
            opt.flat_g.add_(opt._flat_g2()).mul_(0.5)
            if world > 1:
                dist.all_reduce(opt.flat_g)
                opt.flat_g /= world
            ext.adam_step(opt.flat_p, opt.flat_g, opt.m, opt.v, opt.lr,
                          opt.betas[0], opt.betas[1], opt.eps, 0,
                          self.step_buf)
            return (loss0 + loss1) * 0.5, grads_a, grads_b

        snap = (opt.flat_p.clone(), opt.m.clone(), opt.v.clone(),
                self.seed0.clone(), self.seed1.clone())
        side = torch.cuda.Stream(device=dev)
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):
                one_step(static_tabs=False)
        torch.cuda.current_stream().wait_stream(side)

        import gc

        torch.cuda.synchronize()
        gc.collect()
        gc_on = gc.isenabled()
        gc.disable()
        try:
            self.graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.graph):
                self.loss, self._ga, self._gb = one_step(static_tabs=True)
        finally:
            if gc_on:
                gc.enable()
        # now the grad addresses are final: fill the recorded tables once
        self.tab_a.copy_(host_table(self._ga))
        self.tab_b.copy_(host_table(self._gb))
        # undo the warmup's 3 real updates (see snapshot above)
        opt.flat_p.copy_(snap[0])
        opt.m.copy_(snap[1])
        opt.v.copy_(snap[2])
        self.seed0.copy_(snap[3])
        self.seed1.copy_(snap[4])
        self.step_buf.zero_()
        torch.cuda.synchronize()

    def __call__(self, x, y):
        self.x.copy_(x, non_blocking=True)
        self.y.copy_(y, non_blocking=True)
        self.graph.replay()
        return self.loss
