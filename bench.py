"""Driver benchmark contract: flagship two-phase 3D FNO training step.

Measures sec/batch (forward + distributed relative-Lp loss + backward +
Adam step) of the 3D two-phase-flow FNO (BASELINE.md flagship: width 20,
modes (12,12,12,8), batch 1, T_out 30) on a fixed 64^3 global grid, strong
scaling over 1/2/4/8 GPUs of one node (partitions 1x1x1 / 2x1x1 / 2x2x1 /
2x2x2 over the spatial axes), synthetic data, random-init weights, fp32
(the reference's compute dtype; complex64 spectral path).

Launch:
  python bench.py --gpus 1 --steps K --warmup W          (serial)
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N ...       (one rank per GPU, RCCL)

Rank 0 prints exactly one JSON line per the driver contract.
"""

import argparse
import json
import os
import time

import torch

import dfno_amd as dfno
from dfno_amd.partition import init_distributed, is_distributed, world_rank, world_size
from dfno_amd.partition import compute_distribution_info


# Partitions are chosen over the LEADING spatial axes only: the pencil
# construction then gives P_m == P_x (reference dfno.py:88-89 semantics), so
# R1/R4 are identities and only the truncated spectrum (~18 MB global)
# crosses xGMI in R2/R3 — the 553 MB real-activation all-to-alls the
# reference pays per block never happen on this curve.  (The
# reference's flagship (1,1,1,4,1,1) has the same property.)
PARTITIONS = {
    1: (1, 1, 1, 1, 1, 1),
    2: (1, 1, 2, 1, 1, 1),
    4: (1, 1, 2, 2, 1, 1),
    8: (1, 1, 4, 2, 1, 1),
}

# --heavy-comm: partition the TRAILING spatial axis (z) instead, which makes
# P_m != P_x, so R1/R4 are real full-activation all-to-alls over xGMI every
# block (the hard-comms curve VERDICT.md round-1 asked for).
HEAVY_PARTITIONS = {
    1: (1, 1, 1, 1, 1, 1),
    2: (1, 1, 1, 1, 2, 1),
    4: (1, 1, 1, 2, 2, 1),
    8: (1, 1, 2, 2, 2, 1),
}

# flagship config (BASELINE.md / BASELINE.json): 3D two-phase FNO, fixed grid
GLOBAL_SHAPE = [1, 2, 64, 64, 64, 1]   # (batch, channels, x, y, z, t_in)
OUT_T = 30
WIDTH = 20
MODES = (12, 12, 12, 8)
NUM_BLOCKS = 4


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    # 50 steps: long enough (~1.3 s on 1 GPU) for driver-side GPU-busy
    # sampling to catch the run
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--width", type=int, default=WIDTH)
    p.add_argument("--num-blocks", type=int, default=NUM_BLOCKS)
    p.add_argument("--dtype", choices=["fp32", "bf16"], default="fp32",
                   help="compute dtype; fp32 is the headline (the reference's "
                        "dtype), bf16 is the reduced-precision config "
                        "(BASELINE.json config #2: bf16 storage, fp32 math, "
                        "complex64 spectral path)")
    p.add_argument("--spectral-fp8", action="store_true",
                   help="e4m3 storage for the spectral corner weights "
                        "(BASELINE.json config #5); masters stay complex64")
    p.add_argument("--heavy-comm", action="store_true",
                   help="partition the trailing spatial axis: R1/R4 become "
                        "real full-activation all-to-alls")
    p.add_argument("--partition", type=int, nargs="+", default=None,
                   help="explicit 6-dim partition shape (overrides presets)")
    p.add_argument("--allow-fallback", action="store_true",
                   help="skip the all-ops-native assertion (debug only)")
    args = p.parse_args()

    n = args.gpus
    assert n in PARTITIONS, f"--gpus must be one of {sorted(PARTITIONS)}"
    table = HEAVY_PARTITIONS if args.heavy_comm else PARTITIONS
    pshape = tuple(args.partition) if args.partition else table[n]

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    init_distributed()
    if is_distributed():
        assert world_size() == n, f"world size {world_size()} != --gpus {n}"

    use_cuda = torch.cuda.is_available()
    if use_cuda:
        local_rank = int(os.environ.get("LOCAL_RANK", 0))
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
        device = torch.device("cuda", torch.cuda.current_device())
    else:
        device = torch.device("cpu")

    import numpy as _np
    assert int(_np.prod(pshape)) == n, f"partition {pshape} != {n} ranks"
    P_world, P_x, P_0 = dfno.create_standard_partitions(pshape)

    torch.manual_seed(1234 + max(P_x.rank, 0))
    dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    model = dfno.DistributedFNONd(P_x, GLOBAL_SHAPE, OUT_T, args.width, MODES,
                                  num_blocks=args.num_blocks, device=device,
                                  dtype=dtype, spectral_fp8=args.spectral_fp8)
    criterion = dfno.DistributedRelativeLpLoss(P_x)
    from dfno_amd.optim import Adam as FusedAdam
    if torch.cuda.is_available():
        # fail loudly rather than silently timing a torch-eager fallback
        from dfno_amd import _ext
        _ext.get(required=True)
    optimizer = FusedAdam(model.parameters(), lr=1e-3)

    # synthetic local shards of the global tensors
    info_x = compute_distribution_info(P_x, GLOBAL_SHAPE)
    out_shape = [GLOBAL_SHAPE[0], 1, *GLOBAL_SHAPE[2:-1], OUT_T]
    info_y = compute_distribution_info(P_x, out_shape)
    x = torch.rand(*info_x["shape"], device=device, dtype=dtype)
    y_true = torch.rand(*info_y["shape"], device=device, dtype=dtype)

    def step():
        optimizer.zero_grad(set_to_none=True)
        y = model(x)
        loss = criterion(y, y_true)
        loss.backward()
        optimizer.step()
        return loss

    dt_comm_acc = 0.0
    for _ in range(args.warmup):
        step()

    if use_cuda and not args.allow_fallback:
        # the warmup steps must not have left the native kernels anywhere
        # (assert native *usage*, not just that
        # the extension loads)
        from dfno_amd import dispatch
        dispatch.assert_all_native()

    from dfno_amd import timing as _timing
    if use_cuda:
        _timing.enable_event_timing()

    P_x.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(args.steps):
        step()
        dt_comm_acc += model.dt_comm
    if use_cuda:
        torch.cuda.synchronize()
    P_x.barrier()
    t1 = time.time()
    # honest device-side comm time (CUDA events around every collective);
    # dt_comm_acc stays the reference-protocol launch-side host timer
    dt_comm_dev = _timing.device_seconds() if use_cuda else dt_comm_acc
    _timing.disable_event_timing()

    elapsed = t1 - t0
    # max over ranks
    elapsed = P_x.allreduce_scalar(elapsed, op="max")
    sec_per_batch = elapsed / args.steps

    if max(P_x.rank, 0) == 0:
        print(json.dumps({
            "metric": "sec/batch",
            "value": sec_per_batch,
            "unit": "s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": sec_per_batch * 1e3,
            "higher_is_better": False,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic (random input/target shards, random-init weights)",
            "dt_comm_per_step": dt_comm_dev / args.steps,
            "dt_comm_launch_per_step": dt_comm_acc / args.steps,
            "config": {
                "model": "dfno-two-phase-3d",
                "global_batch": GLOBAL_SHAPE[0],
                "grid": GLOBAL_SHAPE[2:5],
                "t_in": GLOBAL_SHAPE[5],
                "t_out": OUT_T,
                "width": args.width,
                "modes": list(MODES),
                "num_blocks": args.num_blocks,
                "parallelism": "spatial-model-parallel " + "x".join(map(str, pshape)),
                "step": "fwd+relative-Lp-loss+bwd+adam",
                "spectral_weights": "fp8-e4m3" if args.spectral_fp8 else "complex64",
            },
        }))


if __name__ == "__main__":
    main()
    dfno.finalize_distributed()
