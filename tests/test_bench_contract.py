"""The driver depends on bench.py's exact output contract: JSON lines on
stdout with the documented schema, the LAST being the inference record. Run
it end-to-end on CPU (tiny step counts; eager torch path; the train section
skips without a GPU) and validate every field the driver reads."""

import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract_cpu():
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"),
         "--steps", "2", "--warmup", "1", "--batch", "8", "--depth", "2"],
        cwd=ROOT, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    # on CPU the train section skips; on GPU there are two records with the
    # inference one LAST (the driver parses the last line)
    assert len(lines) >= 1, out.stdout
    r = json.loads(lines[-1])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in r, key
    assert r["metric"] == "inference_bases_per_sec"
    assert r["n_gpus"] == 1 and r["steps"] == 2 and r["warmup"] == 1
    assert r["higher_is_better"] is True
    assert r["scaling"] == "weak"
    assert r["data"] == "synthetic"
    assert r["value"] > 0 and r["ms_per_step"] > 0
    cfg = r["config"]
    for key in ("model", "global_batch", "seq_len", "parallelism"):
        assert key in cfg, key
    assert cfg["global_batch"] == 8 and cfg["seq_len"] == 90


def test_hw_queue_default_set_on_import():
    """The 4-HW-queue default serializes the serving streams; the package
    must raise it before HIP init (PERF_HISTORY.md)."""
    out = subprocess.run(
        [sys.executable, "-c",
         "import roko_amd, os; print(os.environ['GPU_MAX_HW_QUEUES'])"],
        cwd=ROOT, capture_output=True, text=True, timeout=120,
        env={k: v for k, v in os.environ.items()
             if k != "GPU_MAX_HW_QUEUES"},
    )
    assert out.returncode == 0, out.stderr[-500:]
    assert out.stdout.strip() == "20"


def test_bench_gpus_flag_validated():
    """--gpus N without a matching torchrun WORLD_SIZE must exit loudly
    (round-1 flagged the flag as cosmetic)."""
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--gpus", "4",
         "--steps", "1", "--warmup", "0"],
        cwd=ROOT, capture_output=True, text=True, timeout=300,
    )
    assert out.returncode != 0
    assert "WORLD_SIZE" in out.stderr


def test_bench_world2_gloo():
    """The driver launches bench.py under torchrun with --nproc-per-node N;
    rehearse that exact shape at N=2 on CPU (gloo): rendezvous, per-rank
    work, MAX-reduce of elapsed, and rank-0-only JSON output."""
    env = dict(os.environ)
    env.pop("RANK", None), env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29871", os.path.join(ROOT, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1",
         "--batch", "4", "--depth", "2"],
        cwd=ROOT, capture_output=True, text=True, timeout=900, env=env,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    # rank 0 prints once per record; no duplicate records from rank 1
    recs = [json.loads(l) for l in lines]
    assert len([r for r in recs if r["metric"] == "inference_bases_per_sec"]) == 1
    r = recs[-1]
    assert r["n_gpus"] == 2
    assert r["config"]["parallelism"] == "dp2"
    assert r["config"]["global_batch"] == 8
