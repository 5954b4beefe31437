"""Driver-contract regression tests: bench.py must emit exactly one valid
JSON line with the agreed keys, and the torchrun launch path must work."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    res = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--model",
         "resnet50", "--steps", "1", "--warmup", "0", "--batch", "2"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert res.returncode == 0, res.stderr[-2000:]
    lines = [ln for ln in res.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, res.stdout
    out = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in out, f"missing key {key}"
    assert out["n_gpus"] == 1
    assert out["steps"] == 1
    assert out["unit"] == "img/sec"
    assert out["higher_is_better"] is True
    assert out["scaling"] == "weak"
    assert out["data"] == "synthetic"
    assert isinstance(out["value"], (int, float)) and out["value"] > 0
    assert out["config"]["global_batch"] == 2
    assert out["config"]["parallelism"] == "dp1"


def test_bench_torchrun_world2_cpu():
    """The driver's SCALE launch shape: torchrun --nproc-per-node N
    bench.py --gpus N.  On CPU this exercises gloo init, BucketedDDP
    world=2, the MAX-over-ranks reduction, and the single-JSON-line
    contract from rank 0 only."""
    res = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29534",
         os.path.join(REPO, "bench.py"), "--gpus", "2", "--model",
         "resnet50", "--steps", "1", "--warmup", "0", "--batch", "2"],
        capture_output=True, text=True, timeout=600, cwd=REPO,
    )
    assert res.returncode == 0, res.stderr[-2000:]
    lines = [ln for ln in res.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, res.stdout  # rank 0 only
    out = json.loads(lines[0])
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "dp2"
    assert out["config"]["global_batch"] == 4
    assert out["value"] > 0


def test_allreduce_perf_torchrun_cpu():
    env = dict(os.environ)
    res = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29533",
         os.path.join(REPO, "benchmarks", "allreduce_perf.py"),
         "--max-bytes", "16384", "--iters", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=300, cwd=REPO, env=env,
    )
    assert res.returncode == 0, res.stderr[-2000:]
    summary = [ln for ln in res.stdout.splitlines()
               if ln.startswith('{"bench"')]
    assert summary, res.stdout
    s = json.loads(summary[0])
    assert s["world"] == 2 and s["backend"] == "gloo"


def test_bench_busbw_ab_companion_cpu():
    """--ab always exercises the plugin-vs-stock busbw A/B sub-launch
    machinery (two sub-torchruns, JSON parse, companion embedding) on CPU
    with gloo — the exact path the driver's N>1 GPU run takes."""
    res = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--model",
         "resnet50", "--steps", "1", "--warmup", "0", "--batch", "2",
         "--ab", "always", "--ab-iters", "2", "--ab-max-bytes", "4096"],
        capture_output=True, text=True, timeout=600, cwd=REPO,
    )
    assert res.returncode == 0, res.stderr[-2000:]
    lines = [ln for ln in res.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, res.stdout
    out = json.loads(lines[0])
    ab = out.get("companion_busbw_ab")
    assert ab is not None
    assert ab["ab_status"] == "ok", ab
    assert ab["sizes"][0] == 8
    assert len(ab["busbw_plugin_GBps"]) == len(ab["sizes"])
    assert len(ab["busbw_stock_GBps"]) == len(ab["sizes"])
    assert all(isinstance(x, (int, float)) for x in ab["busbw_plugin_GBps"])
