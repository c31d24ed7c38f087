"""Benchmark harness (driver contract — see project brief).

Measures BOTH halves of the BASELINE.json metric ("inference bases/sec +
train windows/sec at b=128") on N GPUs of one node, by default as two timed
sections of one invocation:

  * train: the fused HIP train step (the same path the roko_amd.train CLI
    runs on GPU) — full forward, backward, FusedAdam update per step;
  * inference: pipelined serving of b=128 windows through the gfx950 kernel
    path, fused argmax, predictions copied to host (bases = windows *
    30-column stride).

  python bench.py --gpus N --steps K --warmup W [--mode both|inference|train]

One JSON line is printed per section; the LAST line is the inference record
(comparable round to round) and it carries the train numbers in its config.
Under torchrun (one rank per GPU, RCCL) inference ranks run independent
batch streams (weak scaling; no collectives — SURVEY.md §2.5); train ranks
all-reduce gradients each step (weak scaling: global batch = 128 * N). Max
elapsed over ranks is used everywhere.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

# must precede HIP runtime init: 4 default HW queues serialize the pipelined
# serving streams (profiles/PERF_HISTORY.md — measured +50% at 16 queues)
os.environ.setdefault("GPU_MAX_HW_QUEUES", "20")

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from roko_amd import config as C
from roko_amd.model import RokoModel
from roko_amd.parallel.ddp import init_distributed


def bench_inference(args, rank, world, device):
    torch.manual_seed(0)
    model = RokoModel().to(device).eval()
    pipe = None
    if device.type == "cuda":
        from roko_amd import ops
        ops.require()
        from roko_amd.ops.forward import InferencePipeline
        pipe = InferencePipeline(model, args.batch, depth=args.depth)
    g = torch.Generator().manual_seed(1234 + rank)
    nbuf = max(4, args.depth + 1)
    xs = [
        torch.randint(0, C.NUM_BASE_IDS, (args.batch, C.WINDOW_ROWS, C.WINDOW_COLS),
                      generator=g, dtype=torch.uint8).to(device)
        for _ in range(nbuf)
    ]

    # One serving "step" = BATCHES_PER_STEP pipelined b=128 forwards. The
    # pipeline holds `depth` batches in flight, so a timed region of only a
    # few single-batch submissions is dominated by fill+drain and
    # under-reports steady-state throughput by ~25-50% (r1's driver record:
    # 14.3M at --steps 20 vs 19.8M sustained). Grouping a fixed quantum of
    # batches per step amortizes the pipeline edges inside the same
    # sync-bracketed timed region — every batch still runs the full model
    # and completes inside the timer; the quantum is reported in config.
    # (32 x 20 steps = 640 batches puts the post-sync refill ramp under
    # ~3% of the measurement; the 120 s soak is the steady-state truth.)
    BATCHES_PER_STEP = 32 if pipe is not None else 1
    if pipe is not None:
        def step(i):
            for j in range(BATCHES_PER_STEP):
                pipe.submit(xs[(i * BATCHES_PER_STEP + j) % nbuf],
                            copy_out=False)
    else:
        def step(i):
            model(xs[i % nbuf].long()).argmax(dim=2).to("cpu")

    for i in range(args.warmup):
        step(i)
    if device.type == "cuda":
        torch.cuda.synchronize()
    if world > 1:
        torch.distributed.barrier()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i)
    if device.type == "cuda":
        torch.cuda.synchronize()
    if world > 1:
        torch.distributed.barrier()
    elapsed = time.perf_counter() - t0

    if world > 1:
        t = torch.tensor([elapsed], device=device if device.type == "cuda" else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    bases = world * args.steps * BATCHES_PER_STEP * args.batch * C.WINDOW_STRIDE
    return {
        "metric": "inference_bases_per_sec",
        "value": bases / elapsed,
        "unit": "bases/s",
        "ms_per_step": elapsed / args.steps * 1000.0,
        "batches_per_step": BATCHES_PER_STEP,
    }


def bench_train(args, rank, world, device):
    from roko_amd.ops.train import (FusedAdam, GraphedTrainStep,
                                    fused_param_order, fused_train_step,
                                    train_step_available)

    if device.type != "cuda" or not train_step_available():
        raise SystemExit("train bench requires the fused HIP train step on GPU")
    torch.manual_seed(0)
    model = RokoModel().to(device).train()
    opt = FusedAdam(fused_param_order(model), lr=C.LR)
    g = torch.Generator().manual_seed(99 + rank)
    x = torch.randint(0, C.NUM_BASE_IDS, (args.batch, C.WINDOW_ROWS, C.WINDOW_COLS),
                      generator=g, dtype=torch.uint8).to(device)
    y = torch.randint(0, C.NUM_CLASSES, (args.batch, C.WINDOW_COLS),
                      generator=g).to(device)

    stepper = None
    # default is the proven eager fused step: the captured variants abort
    # (uncatchable SIGABRT) on some capture-illegal operation inside the
    # multi-stream graph — keep them strictly opt-in experiments
    mode = os.environ.get("ROKO_TRAIN_STEP", "eager")
    if mode in ("dualgraph", "graph"):
        try:
            from roko_amd.ops.train import GraphedDualTrainStep
            cls = GraphedDualTrainStep if mode == "dualgraph" else GraphedTrainStep
            stepper = cls(model, opt, args.batch, world)
        except Exception as e:  # noqa: BLE001 — capture support is optional
            if rank == 0:
                print(f"hipGraph train capture unavailable ({e!r}); "
                      "falling back to eager steps", file=sys.stderr)
    if stepper is not None:
        def fused_train_step(model_, x_, y_, opt_):  # noqa: F811 shadow
            return stepper(x_, y_)
    elif mode == "dual":
        # measured SLOWER than the single-stream fused step (27.6k vs 33.2k
        # windows/s): halving the batch doubles the launch count and the
        # host becomes the critical path before the GRU-overlap win lands
        from roko_amd.ops.train import dual_stream_train_step
        dual_streams = (torch.cuda.Stream(device=device),
                        torch.cuda.Stream(device=device))

        def fused_train_step(model_, x_, y_, opt_):  # noqa: F811 shadow
            return dual_stream_train_step(model_, x_, y_, opt_, dual_streams)

    for _ in range(args.warmup):
        fused_train_step(model, x, y, opt)
    torch.cuda.synchronize()
    if world > 1:
        torch.distributed.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        fused_train_step(model, x, y, opt)
    torch.cuda.synchronize()
    if world > 1:
        torch.distributed.barrier()
    elapsed = time.perf_counter() - t0
    if world > 1:
        t = torch.tensor([elapsed], device=device)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())
    windows = world * args.steps * args.batch
    return {
        "metric": "train_windows_per_sec",
        "value": windows / elapsed,
        "unit": "windows/s",
        "ms_per_step": elapsed / args.steps * 1000.0,
    }


def _record(res, args, world, device, mode, extra=None):
    out = {
        "metric": res["metric"],
        "value": res["value"],
        "unit": res["unit"],
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": res["ms_per_step"],
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16" if device.type == "cuda" else "float32",
        "data": "synthetic",
        "config": {
            "model": "roko bi-GRU polisher (r10-shape: 200x90 windows, "
                     "emb50, fc 200->100->10, GRU 500/128x3 bidir, 5-class)",
            "global_batch": args.batch * world,
            "seq_len": C.WINDOW_COLS,
            "parallelism": f"dp{world}",
            "mode": mode,
            "bases_per_window": C.WINDOW_STRIDE,
        },
    }
    if "batches_per_step" in res:
        out["config"]["batches_per_step"] = res["batches_per_step"]
    if extra:
        out["config"].update(extra)
    return out


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=400)
    p.add_argument("--warmup", type=int, default=50)
    p.add_argument("--batch", type=int, default=C.BATCH_SIZE)
    p.add_argument("--mode", choices=["both", "inference", "train"],
                   default="both",
                   help="'both' (default) measures train windows/s first, "
                        "then inference bases/s — the two halves of the "
                        "BASELINE metric — and prints one JSON line each "
                        "(inference last)")
    p.add_argument("--depth", type=int, default=48,
                   help="in-flight batches / HIP streams (inference mode); kept "
                        "below the default warmup so every hipGraph capture "
                        "happens untimed")
    args = p.parse_args()

    rank, local_rank, world = init_distributed()
    if args.gpus != world:
        raise SystemExit(
            f"--gpus {args.gpus} but WORLD_SIZE is {world}: for N>1 launch "
            "via torchrun --nproc-per-node N (the flag is validated, not a "
            "process launcher)")
    device = (
        torch.device("cuda", local_rank)
        if torch.cuda.is_available()
        else torch.device("cpu")
    )

    from roko_amd.ops.train import train_step_available

    can_train = device.type == "cuda" and train_step_available()
    records = []
    if args.mode == "train":
        res = bench_train(args, rank, world, device)
        records.append(_record(res, args, world, device, "train"))
    elif args.mode == "inference":
        res = bench_inference(args, rank, world, device)
        records.append(_record(res, args, world, device, "inference"))
    else:  # both
        tres = None
        if can_train:
            tres = bench_train(args, rank, world, device)
            records.append(_record(tres, args, world, device, "train"))
        elif rank == 0:
            print("train section skipped: fused HIP step needs a GPU",
                  file=sys.stderr)
        res = bench_inference(args, rank, world, device)
        extra = {}
        if tres is not None:
            extra = {"train_windows_per_sec": tres["value"],
                     "train_ms_per_step": tres["ms_per_step"]}
        records.append(_record(res, args, world, device, "inference", extra))

    if rank == 0:
        for out in records:
            print(json.dumps(out))


if __name__ == "__main__":
    main()
