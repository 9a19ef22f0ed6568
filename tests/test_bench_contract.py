"""The driver depends on bench.py's CLI + JSON contract; pin it on CPU.

Runs the real bench (llama-tiny, 1 step, fp32/CPU) in a subprocess and
validates the single JSON line against the contract the driver parses.
"""

import json
import os
import subprocess
import sys


def test_bench_json_contract(tmp_path):
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proc = subprocess.run(
        [sys.executable, os.path.join(repo, "bench.py"), "--gpus", "1",
         "--model", "llama-tiny", "--steps", "1", "--warmup", "0",
         "--local-samples", "8", "--batch-size", "8"],
        env=env, capture_output=True, text=True, timeout=600,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    lines = [l for l in proc.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"exactly one JSON line expected: {proc.stdout}"
    d = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, f"missing {key}"
    assert d["n_gpus"] == 1 and d["steps"] == 1
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert "global_batch" in d["config"] and "parallelism" in d["config"]


def test_bench_world2_driver_launch_shape():
    """The driver's exact multi-rank form: torch.distributed.run
    --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 bench.py
    --gpus 2. On CPU this runs the gloo data plane with a real 2-rank
    aggregation, per-rank Dirichlet shards, MAX-over-ranks timing, and
    must print exactly ONE JSON line (rank 0) with the whole-job value."""
    env = dict(os.environ)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env["PYTHONPATH"] = repo
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29917", os.path.join(repo, "bench.py"),
         "--gpus", "2", "--model", "llama-tiny", "--steps", "1",
         "--warmup", "0", "--local-samples", "8", "--batch-size", "8"],
        env=env, capture_output=True, text=True, timeout=600,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    lines = [l for l in proc.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"exactly one JSON line expected: {proc.stdout}"
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert d["value"] > 0 and d["ms_per_step"] > 0


def test_bench_refuses_world_mismatch():
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env["WORLD_SIZE"] = "4"  # pretend a rendezvous exists with wrong size
    env["RANK"] = "0"
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proc = subprocess.run(
        [sys.executable, os.path.join(repo, "bench.py"), "--gpus", "2",
         "--model", "llama-tiny", "--steps", "1", "--warmup", "0"],
        env=env, capture_output=True, text=True, timeout=120,
    )
    assert proc.returncode != 0
    assert "refusing" in (proc.stderr + proc.stdout)
