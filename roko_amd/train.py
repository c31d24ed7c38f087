"""Training CLI + engine.

Role-equivalent to the reference's roko/train.py:18-129 (forward ->
cross-entropy -> Adam, val accuracy, EarlyStopping(patience=7), per-epoch
best-by-val-acc ``.pth`` checkpoints), re-built as an explicit engine with
distributed data parallelism:

  * one process per GPU (torchrun), RCCL all-reduce over xGMI
    (roko_amd.parallel.ddp / FusedAdam.allreduce_grads);
  * checkpoints are plain reference-format state_dicts
    (``rnn_model_<epoch>_acc=<acc>.pth`` — SURVEY.md §5.4), so they load in
    either framework;
  * on GPU the DEFAULT step is the fused HIP path benchmarked in bench.py
    (fused front/CE kernels, side-stream weight grads, FusedAdam with one
    flat all-reduce — ops/train.py); ``ROKO_TRAIN_PATH=autograd`` or a batch
    size not divisible by 32 falls back to autograd + torch Adam +
    GradReducer. CPU always runs the autograd reference path.

Usage:
  python -m roko_amd.train <train.rkw|dir> <out_dir> [--val v.rkw] [--memory]
      [--t workers] [--b batch] [--epochs N] [--lr LR] [--patience P]
"""

from __future__ import annotations

import argparse
import math
import os
import time
from typing import Optional

import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader, DistributedSampler

from . import config as C
from .config import TrainConfig
from .datasets import InMemoryTrainDataset, TrainDataset
from .model import RokoModel
from .parallel.ddp import GradReducer, init_distributed
from .utils.metrics import Meter


class EarlyStopper:
    """Stop after `patience` epochs without val-accuracy improvement
    (reference: train.py:74-80)."""

    def __init__(self, patience: int):
        self.patience = patience
        self.best = -math.inf
        self.bad = 0

    def step(self, score: float) -> bool:
        """Returns True when training should stop."""
        if score > self.best:
            self.best = score
            self.bad = 0
            return False
        self.bad += 1
        return self.bad >= self.patience


class CheckpointManager:
    """Keep the best-by-score reference-format state_dicts
    (reference: train.py:82-84 — ignite ModelCheckpoint semantics), plus a
    full resume sidecar (``train_state.pt``: optimizer moments, epoch, RNG,
    early-stop state) — the reference cannot resume at all;
    the ``.pth`` weight files stay pure state_dicts for cross-framework
    compatibility."""

    def __init__(self, out_dir: str, keep: int = 2):
        self.out_dir = out_dir
        self.keep = keep
        self.saved: list[tuple[float, str]] = []
        os.makedirs(out_dir, exist_ok=True)

    def save(self, model: torch.nn.Module, epoch: int, score: float) -> Optional[str]:
        path = os.path.join(self.out_dir, f"rnn_model_{epoch}_acc={score:.4f}.pth")
        if self.saved and score <= min(s for s, _ in self.saved) and len(self.saved) >= self.keep:
            return None
        torch.save(model.state_dict(), path)
        self.saved.append((score, path))
        self.saved.sort(reverse=True)
        while len(self.saved) > self.keep:
            _, drop = self.saved.pop()
            if os.path.exists(drop):
                os.remove(drop)
        return path

    @property
    def state_path(self) -> str:
        return os.path.join(self.out_dir, "train_state.pt")

    def save_state(self, model, opt, epoch: int, stopper: "EarlyStopper") -> None:
        torch.save(
            {
                "model": model.state_dict(),
                "opt": opt.state_dict(),
                "epoch": epoch,
                "stopper": {"best": stopper.best, "bad": stopper.bad},
                "rng": torch.get_rng_state(),
                "cuda_rng": (
                    torch.cuda.get_rng_state()
                    if torch.cuda.is_available() else None
                ),
                "saved": self.saved,
            },
            self.state_path,
        )

    def load_state(self, model, opt, stopper: "EarlyStopper") -> int:
        """Restore a previous run; returns the next epoch to run (1-based)."""
        st = torch.load(self.state_path, map_location="cpu", weights_only=False)
        model.load_state_dict(st["model"])
        opt.load_state_dict(st["opt"])
        stopper.best = st["stopper"]["best"]
        stopper.bad = st["stopper"]["bad"]
        torch.set_rng_state(st["rng"])
        if st.get("cuda_rng") is not None and torch.cuda.is_available():
            torch.cuda.set_rng_state(st["cuda_rng"])
        self.saved = [tuple(x) for x in st.get("saved", [])]
        return int(st["epoch"]) + 1


def evaluate(model, loader, device, rank: int = 0, world: int = 1) -> tuple[float, float]:
    """(accuracy, loss) over a loader (reference: train.py:57-63,69-71).

    With world > 1 the batches are sharded round-robin across ranks and the
    counts all-reduced, so every rank returns the same full-set numbers at
    1/world of the per-rank work (the reference runs the whole set on every
    rank)."""
    import torch.distributed as dist

    model.eval()
    correct, total, loss_sum, batches = 0, 0, 0.0, 0
    with torch.no_grad():
        for i, (x, y) in enumerate(loader):
            if world > 1 and i % world != rank:
                continue
            x, y = x.to(device), y.to(device)
            logits = model(x)
            loss = F.cross_entropy(logits.transpose(1, 2), y)
            pred = logits.argmax(dim=2)
            correct += (pred == y).sum().item()
            total += y.numel()
            loss_sum += loss.item()
            batches += 1
    model.train()
    if world > 1 and dist.is_initialized():
        t = torch.tensor([correct, total, loss_sum, batches],
                         dtype=torch.float64, device=device)
        dist.all_reduce(t)
        correct, total, loss_sum, batches = t.tolist()
    if total == 0:
        return 0.0, 0.0
    return correct / total, loss_sum / max(batches, 1)


def _select_fused_path(device: torch.device, cfg: TrainConfig) -> bool:
    """GPU default = the fused HIP step bench.py measures; autograd only on
    CPU, on request (ROKO_TRAIN_PATH=autograd), or for batch sizes the
    kernels reject (not a multiple of 32)."""
    if device.type != "cuda":
        return False
    if os.environ.get("ROKO_TRAIN_PATH", "fused") == "autograd":
        return False
    if cfg.batch_size % 32 != 0:
        return False
    from .ops.train import train_step_available

    return train_step_available()


def _broadcast_initial_state(model: torch.nn.Module) -> None:
    """Rank-0 weights to all ranks (fused path has no GradReducer)."""
    import torch.distributed as dist

    if dist.is_initialized() and dist.get_world_size() > 1:
        for t in model.state_dict().values():
            if t.is_floating_point() or t.dtype in (torch.int64, torch.int32):
                dist.broadcast(t, src=0)


def train(
    train_path: str,
    out_dir: str,
    val_path: Optional[str] = None,
    cfg: Optional[TrainConfig] = None,
    device: Optional[torch.device] = None,
    log=print,
    max_steps: Optional[int] = None,
):
    cfg = cfg or TrainConfig()
    rank, local_rank, world = init_distributed()
    if device is None:
        device = torch.device("cuda", local_rank) if torch.cuda.is_available() else torch.device("cpu")
    torch.manual_seed(cfg.seed + rank)

    ds_cls = InMemoryTrainDataset if cfg.in_memory else TrainDataset
    train_ds = ds_cls(train_path)
    sampler = (
        DistributedSampler(train_ds, num_replicas=world, rank=rank, seed=cfg.seed)
        if world > 1
        else None
    )
    train_dl = DataLoader(
        train_ds,
        batch_size=cfg.batch_size,
        shuffle=(sampler is None),
        sampler=sampler,
        num_workers=cfg.workers,
        drop_last=True,
        pin_memory=torch.cuda.is_available(),
    )
    val_dl = None
    if val_path:
        val_ds = ds_cls(val_path)
        val_dl = DataLoader(val_ds, batch_size=cfg.batch_size, num_workers=cfg.workers)

    model = RokoModel().to(device)
    use_fused = _select_fused_path(device, cfg)
    if use_fused:
        from .ops.train import FusedAdam, fused_param_order

        # DP sync for the fused path is FusedAdam.allreduce_grads (one flat
        # all-reduce per step) — no GradReducer hooks (the fused backward
        # assigns grads outside autograd's accumulate hooks). The param
        # order makes the GRU direction pairs flat-adjacent so the per-step
        # weight packs are zero-copy views (ops/train.py fused_param_order).
        _broadcast_initial_state(model)
        opt = FusedAdam(fused_param_order(model), lr=cfg.lr)
        reducer = None
    else:
        reducer = GradReducer(list(model.parameters()), cfg.bucket_bytes)
        reducer.sync_module_buffers_and_params(model)
        opt = torch.optim.Adam(model.parameters(), lr=cfg.lr)
    if rank == 0:
        log(f"train path: {'fused HIP step' if use_fused else 'autograd'}")

    stopper = EarlyStopper(cfg.patience)
    ckpt = CheckpointManager(out_dir)
    start_epoch = 1
    if cfg.resume and os.path.exists(ckpt.state_path):
        start_epoch = ckpt.load_state(model, opt, stopper)
        model = model.to(device)
        if rank == 0:
            log(f"resumed from {ckpt.state_path} at epoch {start_epoch}")
    model.train()

    step = 0
    history = []
    meter = Meter("train", rank=rank)
    for epoch in range(start_epoch, cfg.epochs + 1):
        if sampler is not None:
            sampler.set_epoch(epoch)
        t0 = time.time()
        n_batches = 0
        # device-side loss accumulator: a per-step .item() would sync the
        # stream and stall the fused step's side-stream overlap
        loss_acc = torch.zeros((), device=device)
        for x, y in train_dl:
            x = x.to(device, non_blocking=True)
            y = y.to(device, non_blocking=True)
            if use_fused:
                from .ops.train import fused_train_step

                loss = fused_train_step(model, x, y, opt)
            else:
                logits = model(x)
                loss = F.cross_entropy(logits.transpose(1, 2), y)
                opt.zero_grad(set_to_none=False)
                loss.backward()
                reducer.finish()
                opt.step()
            loss_acc += loss.detach()
            n_batches += 1
            step += 1
            meter.add(windows=len(x))
            if max_steps is not None and step >= max_steps:
                break
        if device.type == "cuda":
            torch.cuda.synchronize()
        dt = time.time() - t0
        wps = n_batches * cfg.batch_size * world / max(dt, 1e-9)
        run_loss = float(loss_acc.item())

        if val_dl is not None:
            acc, vloss = evaluate(model, val_dl, device, rank=rank, world=world)
        else:
            acc, vloss = float("nan"), float("nan")
        score = acc if val_dl is not None else -run_loss / max(n_batches, 1)
        history.append(
            {"epoch": epoch, "train_loss": run_loss / max(n_batches, 1),
             "val_acc": acc, "val_loss": vloss, "windows_per_sec": wps}
        )
        if rank == 0:
            log(
                f"epoch {epoch}: loss {run_loss / max(n_batches, 1):.4f} "
                f"val_acc {acc:.4f} ({wps:.0f} windows/s, {dt:.1f}s)"
            )
            ckpt.save(model, epoch, score if not math.isnan(score) else 0.0)
            ckpt.save_state(model, opt, epoch, stopper)
        if max_steps is not None and step >= max_steps:
            break
        if val_dl is not None and stopper.step(score):
            if rank == 0:
                log(f"early stop at epoch {epoch} (no val-acc gain in {cfg.patience})")
            break

    meter.close()
    if reducer is not None:
        reducer.remove()
    return model, history


def main(argv=None):
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("train", help="training .rkw file or directory")
    p.add_argument("out", help="checkpoint output directory")
    p.add_argument("--val", default=None, help="validation .rkw")
    p.add_argument("--memory", action="store_true", help="load dataset to RAM")
    p.add_argument("--t", type=int, default=0, help="DataLoader workers")
    p.add_argument("--b", type=int, default=C.BATCH_SIZE, help="batch size")
    p.add_argument("--epochs", type=int, default=C.EPOCHS)
    p.add_argument("--lr", type=float, default=C.LR)
    p.add_argument("--patience", type=int, default=C.PATIENCE)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--resume", action="store_true",
                   help="resume from <out>/train_state.pt if present")
    a = p.parse_args(argv)
    cfg = TrainConfig(
        batch_size=a.b, epochs=a.epochs, lr=a.lr, patience=a.patience,
        workers=a.t, in_memory=a.memory, seed=a.seed, resume=a.resume,
    )
    train(a.train, a.out, val_path=a.val, cfg=cfg)


if __name__ == "__main__":
    main()
