"""End-to-end CPU tests of the application layer: trainers, dataset,
benchmark harness, script generator (SURVEY.md L5/L6/L7 parity)."""

import json
import os
import subprocess
import sys
from pathlib import Path

import numpy as np
import pytest
import torch

from dist_utils import run_dist

REPO = Path(__file__).resolve().parent.parent


def run_script(args, cwd, nproc=1, timeout=600):
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO)
    env.pop("WORLD_SIZE", None)
    if nproc > 1:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={nproc}", "--master-addr", "127.0.0.1",
               "--master-port", "29531"] + args
    else:
        cmd = [sys.executable] + args
    r = subprocess.run(cmd, cwd=cwd, env=env, capture_output=True, text=True,
                       timeout=timeout)
    assert r.returncode == 0, f"cmd failed:\n{r.stdout[-3000:]}\n{r.stderr[-3000:]}"
    return r


# ---------------------------------------------------------------------------
# sleipner dataset
# ---------------------------------------------------------------------------

def test_sleipner_synthetic_serial():
    import dfno_amd as dfno
    from dfno_amd.data import DistributedSleipnerDataset3D

    _, P_x, _ = dfno.create_standard_partitions((1, 1, 1, 1, 1, 1))
    ds = DistributedSleipnerDataset3D(P_x, [1, 2], shape=(8, 8, 4, 6),
                                      synthetic=True)
    assert len(ds) == 2
    x, y = ds[0]
    assert x.shape == (2, 8, 8, 4, 1)
    assert y.shape == (1, 8, 8, 4, 6)
    assert float(x.min()) >= 0 and float(x.max()) <= 1
    assert float(y.min()) >= 0 and float(y.max()) <= 1


def test_sleipner_local_h5(tmp_path):
    h5py = pytest.importorskip("h5py")
    import dfno_amd as dfno
    from dfno_amd.data import DistributedSleipnerDataset3D

    rng = np.random.RandomState(0)
    with h5py.File(tmp_path / "sample_0001.h5", "w") as f:
        f.create_dataset("permz", data=rng.rand(8, 8, 4).astype(np.float32))
        f.create_dataset("tops", data=rng.rand(8, 8).astype(np.float32))
        f.create_dataset("saturation", data=rng.rand(7, 8, 8, 4).astype(np.float32))

    _, P_x, _ = dfno.create_standard_partitions((1, 1, 1, 1, 1, 1))
    ds = DistributedSleipnerDataset3D(P_x, [1], shape=(8, 8, 4, 6),
                                      root=str(tmp_path))
    x, y = ds[0]
    assert x.shape == (2, 8, 8, 4, 1)
    assert y.shape == (1, 8, 8, 4, 6)


def _sleipner_dist_body(rank, world):
    import dfno_amd as dfno
    from dfno_amd.partition import Partition
    from dfno_amd.data import DistributedSleipnerDataset3D

    P_x = Partition(tuple(range(world)), (1, 1, 1, world, 1, 1))
    ds = DistributedSleipnerDataset3D(P_x, [3], shape=(8, 8, 4, 6),
                                      synthetic=True)
    x, y = ds[0]
    # each rank holds a Y-slab of 8/world
    assert x.shape == (2, 8, 8 // world, 4, 1)
    assert y.shape == (1, 8, 8 // world, 4, 6)
    # normalization used GLOBAL min/max: all-rank max of y must be ~1
    mx = P_x.allreduce_scalar(float(y.max()), "max")
    assert abs(mx - 1.0) < 1e-6


def test_sleipner_distributed_slabs():
    run_dist(_sleipner_dist_body, 2)


# ---------------------------------------------------------------------------
# trainers / harnesses (subprocess smoke tests, tiny shapes, CPU)
# ---------------------------------------------------------------------------

def test_train_two_phase_serial(tmp_path):
    r = run_script([str(REPO / "training/two_phase/train_two_phase.py"),
                    "--data", "synthetic", "--shape", "8", "8", "4", "6",
                    "--width", "4", "--modes", "2", "2", "2", "2",
                    "--num-train", "2", "--num-valid", "1", "--num-epochs", "1",
                    "--checkpoint-interval", "1",
                    "--out-dir", str(tmp_path)], cwd=str(REPO))
    assert (tmp_path / "model_0000.pt").exists()
    assert (tmp_path / "model_0001_0000.pt").exists()
    assert "training finished." in r.stdout


def test_train_two_phase_resume(tmp_path):
    common = [str(REPO / "training/two_phase/train_two_phase.py"),
              "--data", "synthetic", "--shape", "8", "8", "4", "6",
              "--width", "4", "--modes", "2", "2", "2", "2",
              "--num-train", "2", "--num-valid", "1",
              "--checkpoint-interval", "1", "--out-dir", str(tmp_path)]
    run_script(common + ["--num-epochs", "1"], cwd=str(REPO))
    assert (tmp_path / "train_state_0001_0000.pt").exists()
    r = run_script(common + ["--num-epochs", "2", "--resume", str(tmp_path)],
                   cwd=str(REPO))
    # resumed run starts at epoch 1 (0-indexed) and finishes epoch 2
    assert "resumed epoch 1" in r.stdout
    assert "epoch = 1, batch = 0" in r.stdout
    assert "epoch = 0, batch" not in r.stdout
    assert (tmp_path / "model_0002_0000.pt").exists()


def test_train_two_phase_resume_missing_state(tmp_path):
    # --resume against a dir with no train_state files must fail loudly
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, str(REPO / "training/two_phase/train_two_phase.py"),
         "--data", "synthetic", "--shape", "8", "8", "4", "6",
         "--width", "4", "--modes", "2", "2", "2", "2",
         "--num-train", "2", "--num-valid", "1", "--num-epochs", "1",
         "--resume", str(tmp_path), "--out-dir", str(tmp_path)],
        cwd=str(REPO), env=env, capture_output=True, text=True, timeout=300)
    assert r.returncode != 0
    assert "train_state" in (r.stderr + r.stdout)


def test_finalize_distributed_idempotent():
    import dfno_amd

    # serial: both calls are no-ops and must not raise
    dfno_amd.finalize_distributed()
    dfno_amd.finalize_distributed()


def test_train_and_test_two_phase_2rank(tmp_path):
    run_script([str(REPO / "training/two_phase/train_two_phase.py"),
                "--data", "synthetic", "--shape", "8", "8", "4", "6",
                "--width", "4", "--modes", "2", "2", "2", "2",
                "--num-train", "2", "--num-valid", "1", "--num-epochs", "1",
                "--checkpoint-interval", "1",
                "--out-dir", str(tmp_path)], cwd=str(REPO), nproc=2)
    # per-rank sharded checkpoints (reference layout)
    assert (tmp_path / "model_0000.pt").exists()
    assert (tmp_path / "model_0001.pt").exists()

    r = run_script([str(REPO / "training/two_phase/test_two_phase.py"),
                    "--data", "synthetic", "--shape", "8", "8", "4", "6",
                    "--width", "4", "--modes", "2", "2", "2", "2",
                    "--model-dir", str(tmp_path),
                    "--out-dir", str(tmp_path)], cwd=str(REPO), nproc=2)
    assert "relative L2 error" in r.stdout
    assert (tmp_path / "prediction.h5").exists() or (tmp_path / "prediction.npz").exists()


def test_navier_stokes_serial(tmp_path):
    run_script([str(REPO / "training/navier_stokes/experiment_navier_stokes.py"),
                "--synthetic", "--partition-shape", "1", "1", "1", "1", "1",
                "--num-data", "4", "--num-epochs", "1", "--batch-size", "2",
                "--grid", "8", "--in-timesteps", "3", "--out-timesteps", "4",
                "--width", "4", "--modes", "2", "2", "2", "--num-blocks", "1",
                "--checkpoint-interval", "1"], cwd=str(tmp_path))
    outs = list((tmp_path / "data").glob("synthetic_*/model_0001_0000.pt"))
    assert outs, "NS checkpoint not written"


def test_bench_harness_serial(tmp_path):
    r = run_script([str(REPO / "benchmarks/bench.py"),
                    "--input-shape", "1", "1", "8", "8", "3",
                    "--partition_shape", "1", "1", "1", "1", "1",
                    "--modes", "2", "2", "2", "--num-timesteps", "4",
                    "--width", "4", "--device", "cpu",
                    "--benchmark-type", "grad",
                    "--output-dir", str(tmp_path)], cwd=str(tmp_path))
    outs = list(tmp_path.glob("*-grad-0-1.json"))
    assert outs
    data = json.loads(outs[0].read_text())
    for k in ("dt", "dt_comm", "dt_comp", "dt_grad"):
        assert k in data


def test_gen_scripts(tmp_path):
    run_script([str(REPO / "benchmarks/gen_scripts.py"), "--max-workers", "8",
                "--local-shape", "1", "1", "16", "16", "16", "10",
                "--modes", "2", "2", "2", "2"], cwd=str(tmp_path))
    for n in ("eval_weak_scaling_spatial_gpu.sh", "grad_weak_scaling_temporal_gpu.sh"):
        assert (tmp_path / n).exists()
        body = (tmp_path / n).read_text()
        assert "torch.distributed.run" in body


def test_bench_harness_2rank(tmp_path):
    # the reference-protocol harness under torchrun (gloo on CPU), the same
    # launch pattern the driver uses for the multi-GPU scaling runs
    r = run_script([str(REPO / "benchmarks/bench.py"),
                    "--input-shape", "1", "1", "8", "8", "3",
                    "--partition_shape", "1", "1", "2", "1", "1",
                    "--modes", "2", "2", "2", "--num-timesteps", "4",
                    "--width", "4", "--device", "cpu",
                    "--benchmark-type", "grad",
                    "--output-dir", str(tmp_path)], cwd=str(tmp_path), nproc=2)
    outs = sorted(tmp_path.glob("*-grad-*-2.json"))
    assert len(outs) == 2, [p.name for p in outs]
    for o in outs:
        d = json.loads(o.read_text())
        assert "dt" in d and "dt_grad" in d
