"""Full-model GPU forward through the hand-written gfx950 kernels.

Orchestration: the fused embed+MLP kernel produces the GRU
input sequence; per layer, the input projections of BOTH directions and all
90 steps are ONE hipBLASLt GEMM (plain GEMM -> library; the brief's rule) and
the recurrence is one persistent kernel launch; the 5-class head is a fused
GEMV(+argmax) kernel. No per-step launches, no MIOpen RNN, no eager fallback.
"""

from __future__ import annotations

import os

import torch

from .. import config as C

#: eval front kernel: the one-hot embed+MLP (factors the read reduction
#: through the 12 base classes — ~2.3x fewer MFMAs per column) does ~1.45x
#: less chip-work than the shared train-front kernel; v3 (wave-private
#: columns, no per-phase barriers) is the measured-fastest default
#: (106.7 us vs v2's 152.4 at b=128). ROKO_FRONT=v2/v1/shared for A/B.
_FRONT = os.environ.get("ROKO_FRONT", "v3")


def _front_eval(ext, ids, w):
    if _FRONT == "shared":
        return ext.front_fwd(ids, w["w1"], w["b1"], w["w2"], w["b2"],
                             w["emb"], 0, 1.0)
    if _FRONT == "v1" or not hasattr(ext, "embed_mlp_fwd2"):
        return ext.embed_mlp_fwd(ids, w["w1"], w["b1"], w["w2"], w["b2"],
                                 w["emb"])
    if _FRONT == "v3" and hasattr(ext, "embed_mlp_fwd3"):
        return ext.embed_mlp_fwd3(ids, w["w1g"], w["b1"], w["w2"], w["b2"],
                                  w["emb"])
    return ext.embed_mlp_fwd2(ids, w["w1g"], w["b1"], w["w2"], w["b2"],
                              w["emb"])


def _ext():
    from . import ext

    return ext()


def _bf16_weights(model) -> dict:
    """Per-model cache of bf16/fp32 kernel-ready weight tensors."""
    params = [
        model.embedding.weight, model.fc1.weight, model.fc1.bias,
        model.fc2.weight, model.fc2.bias, model.fc4.weight, model.fc4.bias,
    ]
    for l in range(C.NUM_LAYERS):
        for suffix in ("", "_reverse"):
            for kind in ("weight_ih", "weight_hh", "bias_ih", "bias_hh"):
                params.append(getattr(model.gru, f"{kind}_l{l}{suffix}"))
    ver = tuple(p._version for p in params) + (params[0].device,)
    cache = getattr(model, "_hip_weight_cache", None)
    if cache is not None and cache["ver"] == ver:
        return cache

    g = model.gru
    c = {"ver": ver}
    c["w1"] = model.fc1.weight.detach().to(torch.bfloat16).contiguous()
    # (112, 232) zero-padded W1 for the chunked v2 front (L2-resident A tile)
    w1g = torch.zeros(112, 232, dtype=torch.bfloat16,
                      device=model.fc1.weight.device)
    w1g[:100, :200] = c["w1"]
    c["w1g"] = w1g
    c["b1"] = model.fc1.bias.detach().float().contiguous()
    c["w2"] = model.fc2.weight.detach().to(torch.bfloat16).contiguous()
    c["b2"] = model.fc2.bias.detach().float().contiguous()
    c["emb"] = model.embedding.weight.detach().to(torch.bfloat16).contiguous()
    c["w4"] = model.fc4.weight.detach().to(torch.bfloat16).contiguous()
    c["b4"] = model.fc4.bias.detach().float().contiguous()
    for l in range(C.NUM_LAYERS):
        wf = getattr(g, f"weight_ih_l{l}").detach()
        wr = getattr(g, f"weight_ih_l{l}_reverse").detach()
        bf = getattr(g, f"bias_ih_l{l}").detach()
        br = getattr(g, f"bias_ih_l{l}_reverse").detach()
        # (in, 768) so xg = x @ w_ih_t is one GEMM covering both directions
        w_cat = torch.cat([wf, wr], dim=0).to(torch.bfloat16)
        c[f"w_ih_t{l}"] = w_cat.t().contiguous()
        # (768, KP) row-padded copy for the in-kernel xg GEMM (KP multiple of
        # 32 and 16-byte-aligned rows; zero pad cols are multiplied by the
        # zero-padded x stage)
        kin = w_cat.shape[1]
        kp = 512 if kin > 256 else 256
        wp = torch.zeros(768, kp, dtype=torch.bfloat16, device=w_cat.device)
        wp[:, :kin] = w_cat
        c[f"w_ih_p{l}"] = wp.contiguous()
        c[f"b_ih{l}"] = torch.cat([bf, br]).to(torch.bfloat16).contiguous()
        c[f"u{l}"] = torch.stack(
            [getattr(g, f"weight_hh_l{l}").detach(),
             getattr(g, f"weight_hh_l{l}_reverse").detach()]
        ).to(torch.bfloat16).contiguous()
        c[f"bhh{l}"] = torch.stack(
            [getattr(g, f"bias_hh_l{l}").detach(),
             getattr(g, f"bias_hh_l{l}_reverse").detach()]
        ).float().contiguous()
    model._hip_weight_cache = c
    return c


def roko_forward(model, x: torch.Tensor) -> torch.Tensor:
    """ids (B, 200, 90) int/uint8 on GPU -> logits (B, 90, 5) fp32."""
    ext = _ext()
    w = _bf16_weights(model)
    B0 = x.shape[0]
    ids = x.to(torch.uint8).contiguous()
    pad = (-B0) % 32
    if pad:
        ids = torch.cat([ids, ids.new_zeros(pad, *ids.shape[1:])])
    B = ids.shape[0]
    T = C.WINDOW_COLS

    seq = _front_eval(ext, ids, w)
    for l in range(C.NUM_LAYERS):
        xg = torch.addmm(
            w[f"b_ih{l}"], seq.reshape(T * B, -1), w[f"w_ih_t{l}"]
        ).view(T, B, 2, 384)
        (hseq,) = ext.gru_layer_fwd(xg.contiguous(), w[f"u{l}"], w[f"bhh{l}"], False)
        seq = hseq.view(T, B, 2 * C.HIDDEN_SIZE)
    (logits,) = ext.head_fwd(seq, w["w4"], w["b4"], True, False)
    return logits[:B0]


class InferencePipeline:
    """Serving-style pipelined inference on one GPU.

    The GRU recurrence is latency-bound: at b=128 one forward occupies only
    8 workgroups of the 256 CUs (profiles/ kernel stats — gru_layer_fwd
    dominates chain time at ~3% chip occupancy). Independent batches
    therefore overlap, so the engine keeps `depth` in-flight batches on
    per-slot HIP streams. Each slot's whole forward (front kernel -> 3x
    (xg GEMM + persistent GRU kernel) -> fused head+argmax -> pinned D2H)
    is enqueued by ONE C++ call (`ext.ServeSlot.submit`); for batches
    >= 256 the slot auto-captures its kernel sequence into a RAW hipGraph
    (BASELINE.json config 4's "hipGraph-captured GRU steps") — a measured
    win at large batch and a measured loss at b=128, both A/B'd
    (profiles/PERF_HISTORY.md; ROKO_GSLOT overrides). The older
    torch-level graph backend is kept as ROKO_SERVE=graph for A/B.

    Every submitted batch runs the full model; nothing is cached or skipped
    — pipelining only overlaps independent batches, as a serving deployment
    would.
    """

    def __init__(self, model, batch: int, depth: int = 4,
                 use_graphs: bool = True):
        from . import require

        require()
        self.batch = batch
        self.depth = depth
        self.model = model
        self.w = _bf16_weights(model)
        self.use_graphs = use_graphs
        dev = next(model.parameters()).device
        self.slots = []
        for _ in range(depth):
            self.slots.append(_Slot(self, batch, dev))
        self._next = 0

    def _forward_amax(self, ids_u8: torch.Tensor) -> torch.Tensor:
        """Static-shape argmax forward (body of roko_argmax, no padding)."""
        ext = _ext()
        w = self.w
        B = ids_u8.shape[0]
        T = C.WINDOW_COLS
        seq = _front_eval(ext, ids_u8, w)
        for l in range(C.NUM_LAYERS):
            xg = torch.addmm(
                w[f"b_ih{l}"], seq.reshape(T * B, -1), w[f"w_ih_t{l}"]
            ).view(T, B, 2, 384)
            (hseq,) = ext.gru_layer_fwd(
                xg.contiguous(), w[f"u{l}"], w[f"bhh{l}"], False
            )
            seq = hseq.view(T, B, 2 * C.HIDDEN_SIZE)
        (amax,) = ext.head_fwd(seq, w["w4"], w["b4"], False, True)
        return amax

    def submit(self, x: torch.Tensor, copy_out: bool = True):
        """Enqueue one batch (device or host tensor, (<=batch, R, W) int).
        Returns a ticket; call ticket() to wait and get predictions
        (n, W) uint8 on pinned host memory (a view — copy before the slot
        is reused `depth` submissions later, or pass copy_out=True)."""
        slot = self.slots[self._next % self.depth]
        self._next += 1
        return slot.submit(x, copy_out)


class _Ticket:
    """Handle for one submitted batch; call it to wait and get the (n, W)
    uint8 predictions. Snapshotted automatically when its slot is reused."""

    def __init__(self, slot: "_Slot", n: int):
        self.slot = slot
        self.n = n
        self.data = None

    def materialize(self):
        if self.data is None:
            self.slot.wait_done()
            self.data = self.slot.host_out[: self.n].clone()

    def __call__(self) -> torch.Tensor:
        self.materialize()
        return self.data


class _Slot:
    """One pipeline slot. Two serving backends:

    * C++ fast path (default): an `ext.ServeSlot` owns the slot stream and
      enqueues the whole forward (input copy, front kernel, 3x xg GEMM + GRU,
      fused head+argmax, pinned D2H) in ONE pybind call. The server is
      host-bound — a torch hipGraph replay costs ~51 us + ~15 us/node of
      host time on ROCm and threads don't scale (runtime-global enqueue
      lock) — so per-batch Python work IS the throughput limit
      (profiles/PERF_HISTORY.md).
    * hipGraph path (ROKO_SERVE=graph): each slot's forward captured once
      and replayed per batch; kept for A/B and as the reference backend.
    """

    def __init__(self, pipe: "InferencePipeline", batch: int, dev):
        self.pipe = pipe
        self.pending = None
        self.host_out = torch.empty(
            (batch, C.WINDOW_COLS), dtype=torch.uint8, pin_memory=True
        )
        self.cpp = None
        self.graph = None
        ext = _ext()
        if pipe.use_graphs and os.environ.get("ROKO_SERVE") != "graph" \
                and hasattr(ext, "ServeSlot"):
            self.cpp = ext.ServeSlot(self.pipe.w, batch, self.host_out)
            # one warm-up pass primes the hipBLASLt workspace/algo cache for
            # the slot stream so submit() never allocates
            warm = torch.zeros((batch, C.WINDOW_ROWS, C.WINDOW_COLS),
                               dtype=torch.uint8, device=dev)
            self.cpp.run(warm, batch)
            self.cpp.sync()
            return
        self.stream = torch.cuda.Stream(device=dev)
        self.x = torch.zeros(
            (batch, C.WINDOW_ROWS, C.WINDOW_COLS), dtype=torch.uint8,
            device=dev
        )
        self.event = torch.cuda.Event()
        with torch.cuda.stream(self.stream):
            self.amax = pipe._forward_amax(self.x)  # warm-up + output buffer
            self.host_out.copy_(self.amax, non_blocking=True)
        self.stream.synchronize()
        if pipe.use_graphs:
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, stream=self.stream):
                self.amax = pipe._forward_amax(self.x)
                # pinned D2H is capture-legal: folding it into the graph
                # saves one host call per submitted batch
                self.host_out.copy_(self.amax, non_blocking=True)
            self.graph = g

    def wait_done(self):
        if self.cpp is not None:
            self.cpp.sync()
        else:
            self.event.synchronize()

    def submit(self, x: torch.Tensor, copy_out: bool):
        n = x.shape[0]
        # preserve the previous tenant's predictions before the slot buffers
        # are overwritten (its ticket may be called arbitrarily late)
        if self.pending is not None:
            self.pending.materialize()
            self.pending = None
        if x.dtype != torch.uint8:
            x = x.to(torch.uint8)
        if self.cpp is not None:
            # ONE C++ call enqueues the whole forward (GIL released). No
            # event sync in the hot path: stream ordering makes buffer
            # reuse safe, tickets sync in materialize(), and
            # hipEventSynchronize itself measured ~80 us with ~32 live
            # streams (profiles/PERF_HISTORY.md)
            self.cpp.submit(x if x.is_contiguous() else x.contiguous(), n)
            t = _Ticket(self, n)
            if copy_out:
                self.pending = t
            return t
        self.event.synchronize()  # previous tenant fully drained
        # the input may have been produced on another stream (usually the
        # default one): order the slot stream behind it before copying
        self.stream.wait_stream(torch.cuda.current_stream())
        if x.is_cuda:
            # x may be the dtype/contiguity temp created above, freed as soon
            # as submit() returns — keep its memory until the copy runs
            x.record_stream(self.stream)
        with torch.cuda.stream(self.stream):
            self.x[:n].copy_(x, non_blocking=True)
            if self.graph is not None:
                self.graph.replay()  # forward + pinned D2H
            else:
                self.amax = self.pipe._forward_amax(self.x)
                self.host_out.copy_(self.amax, non_blocking=True)
            self.event.record(self.stream)
        t = _Ticket(self, n)
        if copy_out:
            self.pending = t
        return t


def roko_argmax(model, x: torch.Tensor) -> torch.Tensor:
    """ids -> per-position class predictions (B, 90) uint8, argmax fused."""
    ext = _ext()
    w = _bf16_weights(model)
    B0 = x.shape[0]
    ids = x.to(torch.uint8).contiguous()
    pad = (-B0) % 32
    if pad:
        ids = torch.cat([ids, ids.new_zeros(pad, *ids.shape[1:])])
    B = ids.shape[0]
    T = C.WINDOW_COLS
    seq = _front_eval(ext, ids, w)
    for l in range(C.NUM_LAYERS):
        xg = torch.addmm(
            w[f"b_ih{l}"], seq.reshape(T * B, -1), w[f"w_ih_t{l}"]
        ).view(T, B, 2, 384)
        (hseq,) = ext.gru_layer_fwd(xg.contiguous(), w[f"u{l}"], w[f"bhh{l}"], False)
        seq = hseq.view(T, B, 2 * C.HIDDEN_SIZE)
    (amax,) = ext.head_fwd(seq, w["w4"], w["b4"], False, True)
    return amax[:B0]
