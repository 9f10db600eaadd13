"""The driver depends on bench.py's CLI and JSON contract; validate it."""

import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_bench_json_contract():
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--gpus", "1", "--steps", "1",
         "--warmup", "0", "--width", "4", "--num-blocks", "1"],
        capture_output=True, text=True, timeout=900, env=env, cwd=str(REPO))
    assert r.returncode == 0, r.stderr[-2000:]
    line = r.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, f"missing {key}"
    assert d["metric"] == "sec/batch"
    assert d["higher_is_better"] is False
    assert d["scaling"] == "strong"
    assert d["n_gpus"] == 1
    assert d["value"] > 0
    assert abs(d["ms_per_step"] - d["value"] * 1e3) < 1e-6
    assert "synthetic" in d["data"]
    for k in ("model", "global_batch", "parallelism"):
        assert k in d["config"]


def test_bench_json_contract_4rank():
    # the driver launches N>1 exactly like this (torchrun, one rank per GPU);
    # on CPU the world runs over gloo with the same (1,1,2,2,1,1) partition
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=4", "--master-addr", "127.0.0.1",
         "--master-port", "29537", str(REPO / "bench.py"), "--gpus", "4",
         "--steps", "1", "--warmup", "0", "--width", "4", "--num-blocks", "1"],
        capture_output=True, text=True, timeout=900, env=env, cwd=str(REPO))
    assert r.returncode == 0, r.stderr[-2000:]
    line = r.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    assert d["n_gpus"] == 4
    assert "2x2" in d["config"]["parallelism"]
    assert d["value"] > 0


def test_bench_json_contract_8rank_driver_partition():
    # the exact launch+partition the driver's 8-GPU SCALE run uses
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=8", "--master-addr", "127.0.0.1",
         "--master-port", "29538", str(REPO / "bench.py"), "--gpus", "8",
         "--steps", "1", "--warmup", "0", "--width", "4", "--num-blocks", "1"],
        capture_output=True, text=True, timeout=900, env=env, cwd=str(REPO))
    assert r.returncode == 0, r.stderr[-2000:]
    d = json.loads(r.stdout.strip().splitlines()[-1])
    assert d["n_gpus"] == 8
    assert "4x2" in d["config"]["parallelism"]
    assert d["value"] > 0


def test_bench_json_contract_8rank_heavy_comm():
    # trailing-axis partition: real R1/R4 all-to-alls + chunk pipeline at 8 ranks
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=8", "--master-addr", "127.0.0.1",
         "--master-port", "29539", str(REPO / "bench.py"), "--gpus", "8",
         "--heavy-comm", "--steps", "1", "--warmup", "0", "--width", "4",
         "--num-blocks", "1"],
        capture_output=True, text=True, timeout=900, env=env, cwd=str(REPO))
    assert r.returncode == 0, r.stderr[-2000:]
    d = json.loads(r.stdout.strip().splitlines()[-1])
    assert d["n_gpus"] == 8
    assert "2x2x2" in d["config"]["parallelism"]
    assert d["value"] > 0
