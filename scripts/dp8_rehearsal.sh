#!/bin/bash
# DP-8 rehearsal: the exact command sequence an 8-GPU
# lease (or the driver's SCALE run) executes, so multi-GPU works first try.
# Usage: scripts/dp8_rehearsal.sh [NGPUS]
#
# Covers, in order:
#   1. RCCL sanity: tiny all-reduce across N ranks (isolates comm stack
#      issues from framework issues)
#   2. bench.py --mode both at N ranks: DP train (FusedAdam flat
#      all-reduce per step) + weak-scaling inference
#   3. torchrun roko_amd.train on synthetic windows: the CLI path with
#      DistributedSampler + sharded validation
set -euo pipefail
cd "$(dirname "$0")/.."
N=${1:-8}
export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}
export GPU_MAX_HW_QUEUES=${GPU_MAX_HW_QUEUES:-20}
TR="python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
    --master-addr 127.0.0.1 --master-port 29641"

echo "== 1/3 RCCL all-reduce sanity ($N ranks) =="
$TR --no-python python -c "
import os, torch, torch.distributed as dist
dist.init_process_group('nccl')
r = dist.get_rank(); torch.cuda.set_device(int(os.environ['LOCAL_RANK']))
t = torch.full((1024,), float(r + 1), device='cuda')
dist.all_reduce(t)
expect = sum(range(1, dist.get_world_size() + 1))
assert torch.allclose(t, torch.full_like(t, float(expect))), t[0]
if r == 0: print('RCCL all-reduce OK, world', dist.get_world_size())
dist.destroy_process_group()
"

echo "== 2/3 bench.py both modes, $N ranks =="
$TR bench.py --gpus "$N" --steps 100 --warmup 20

echo "== 3/3 train CLI, $N ranks, synthetic windows =="
python - <<'EOF'
import numpy as np
from roko_amd import config as C
from roko_amd.rkdata import RkwWriter
rng = np.random.default_rng(0)
w = RkwWriter("/tmp/dp8_train.rkw", inference=False)
n = 2048
P = np.zeros((n, C.WINDOW_COLS, 2), dtype=np.int32)
P[..., 0] = np.arange(C.WINDOW_COLS)[None, :]
w.store("c1", 0, C.WINDOW_COLS,
        P, rng.integers(0, 12, (n, C.WINDOW_ROWS, C.WINDOW_COLS), dtype=np.uint8),
        rng.integers(0, 5, (n, C.WINDOW_COLS), dtype=np.uint8))
w.write_contigs([("c1", "A" * 200)])
w.close()
EOF
$TR -m roko_amd.train /tmp/dp8_train.rkw /tmp/dp8_ckpt --epochs 2 --b 128 --memory
echo "DP-$N rehearsal OK"
