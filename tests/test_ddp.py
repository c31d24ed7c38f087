"""Multi-process data-parallel correctness over gloo (CPU stand-in for RCCL;
same code path, SURVEY.md §5.8 / test strategy (e))."""

import multiprocessing as mp
import os
import pickle

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.nn.functional as F

from roko_amd import config as C
from roko_amd.model import RokoModel
from roko_amd.parallel.ddp import GradReducer


def _worker(rank, world, port, tmpdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(0)  # identical init on every rank

    model = RokoModel()
    model.eval()  # disable dropout for determinism
    reducer = GradReducer(list(model.parameters()), bucket_bytes=1 << 20)

    g = torch.Generator().manual_seed(1)
    x_all = torch.randint(0, 12, (4, C.WINDOW_ROWS, C.WINDOW_COLS), generator=g)
    y_all = torch.randint(0, 5, (4, C.WINDOW_COLS), generator=g)
    # each rank takes half the global batch
    x = x_all[rank * 2 : rank * 2 + 2]
    y = y_all[rank * 2 : rank * 2 + 2]

    logits = model(x)
    loss = F.cross_entropy(logits.transpose(1, 2), y)
    loss.backward()
    reducer.finish()

    if rank == 0:
        grads = {n: p.grad.clone() for n, p in model.named_parameters()}
        with open(os.path.join(tmpdir, "ddp_grads.pkl"), "wb") as f:
            pickle.dump(grads, f)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_grads_match_single_process(tmp_path):
    world = 2
    ctx = mp.get_context("spawn")
    port = 29531
    procs = [
        ctx.Process(target=_worker, args=(r, world, port, str(tmp_path)))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0

    # single-process reference on the full global batch
    torch.manual_seed(0)
    model = RokoModel()
    model.eval()
    g = torch.Generator().manual_seed(1)
    x = torch.randint(0, 12, (4, C.WINDOW_ROWS, C.WINDOW_COLS), generator=g)
    y = torch.randint(0, 5, (4, C.WINDOW_COLS), generator=g)
    logits = model(x)
    loss = F.cross_entropy(logits.transpose(1, 2), y)
    loss.backward()

    import pickle as pkl
    with open(tmp_path / "ddp_grads.pkl", "rb") as f:
        ddp_grads = pkl.load(f)
    for n, p in model.named_parameters():
        # mean of per-rank CE losses == CE of the global batch here because
        # both ranks hold equal-sized batches and CE averages over elements
        assert torch.allclose(ddp_grads[n], p.grad, atol=1e-6), n
