"""Generate weak-scaling launch scripts for the MI355X node.

Counterpart of /root/reference/benchmarks/gen_scripts.py, targeting a single
8-GPU MI355X node launched with torchrun over RCCL (instead of Summit jsrun /
Perlmutter srun over MPI).  Two weak-scaling modes (reference :44-52):

* spatial:  local grid fixed per GPU; global extents (and spatial modes)
  multiply with the partition factors.
* temporal: spatial extents fixed; nt and the temporal mode count multiply
  with the world size.

Validity checks keep the decomposition legal, including the rfft
half-spectrum bound on the trailing dim (reference :55-63).

Usage: python benchmarks/gen_scripts.py [--max-workers 8]
Creates {eval,grad}_weak_scaling_{spatial,temporal}_gpu.sh; run e.g.
  ./grad_weak_scaling_spatial_gpu.sh 4
"""

import os
from argparse import ArgumentParser
from pathlib import Path

import numpy as np

parser = ArgumentParser()
parser.add_argument("--max-workers", "-mw", type=int, default=8)
parser.add_argument("--local-shape", type=int, nargs=6,
                    default=(1, 1, 64, 64, 64, 32),
                    help="per-GPU local shape (b c x y z t)")
parser.add_argument("--modes", type=int, nargs=4, default=(4, 4, 4, 4))
parser.add_argument("--width", type=int, default=20)
args = parser.parse_args()

# 8 GPUs of one MI355X node (xGMI point-to-point)
RUNS = [
    (1, (1, 1, 1, 1, 1, 1)),
    (2, (1, 1, 2, 1, 1, 1)),
    (4, (1, 1, 2, 2, 1, 1)),
    (8, (1, 1, 2, 2, 2, 1)),
]


def format_runs(runs, data_dir, shape, modes, run_type, mode="spatial"):
    out = "#!/bin/bash\nset -x\n\n"
    out += f"data_dir={data_dir}\n"
    out += ('if test "x$1" = x; then\n'
            '  echo "Usage: $0 <numranks>"\n'
            "  exit 0\n"
            "fi\n"
            "ranks=$1\n")

    for nprocs, pshape in runs:
        if mode == "spatial":
            shape_np = [s * ps for s, ps in zip(shape, pshape[:-1])]
            shape_np.append(1)
            modes_np = [m * ps for m, ps in zip(modes, pshape[2:])]
            nt = shape[-1] * pshape[-1]
        else:
            shape_np = [*shape[:-1], 1]
            modes_np = [*modes[:-1], int(np.prod(pshape)) * modes[-1]]
            nt = int(np.prod(pshape)) * shape[-1]

        shape_in = [*shape_np[:-1], nt]
        p1 = pshape[2] * pshape[4]
        p2 = pshape[3] * pshape[5]
        for d, p in ((2, p1), (3, p2), (4, p1)):
            if p > shape_in[d]:
                raise Exception(f"invalid config: partition {pshape} x shape "
                                f"{shape_in} gives a zero-size dim {d}")
        if p2 > shape_in[5] // 2:
            raise Exception(f"invalid config: partition {pshape} exceeds the "
                            f"rfft half-spectrum of dim 5 ({shape_in[5] // 2})")

        launcher = (f"python -m torch.distributed.run --nnodes=1 "
                    f"--nproc-per-node {nprocs} --master-addr 127.0.0.1 "
                    f"--master-port 29517 ")
        out += (f"[[ $ranks -eq '{nprocs}' ]] && {launcher}"
                f"../benchmarks/bench.py --input-shape {' '.join(map(str, shape_np))} "
                f"--modes {' '.join(map(str, modes_np))} "
                f"--partition_shape {' '.join(map(str, pshape))} "
                f"--width {args.width} --num-timesteps {nt} --device cuda "
                f"--benchmark-type {run_type} --output-dir $data_dir\n")
    return out


def create_runscript(name, runs, shape, modes, run_type, mode="spatial"):
    fname = Path(f"{name}.sh")
    with open(fname, "w") as f:
        f.write(format_runs(runs, Path(name), shape, modes, run_type, mode=mode))
    os.chmod(fname, 0o755)
    print(f"created script: {fname.name}")


runs = [r for r in RUNS if r[0] <= args.max_workers]
for name, runtype, mode in [
    ("eval_weak_scaling_spatial_gpu", "eval", "spatial"),
    ("grad_weak_scaling_spatial_gpu", "grad", "spatial"),
    ("eval_weak_scaling_temporal_gpu", "eval", "temporal"),
    ("grad_weak_scaling_temporal_gpu", "grad", "temporal"),
]:
    create_runscript(name, runs, list(args.local_shape), list(args.modes),
                     runtype, mode=mode)


# ---------------------------------------------------------------------------
# 256^3 8-GPU model-parallel capability point (BASELINE.json config #4):
# fixed 256^3 global grid, width 20, modes (16,16,16,8), nt 32, partition
# (1,1,2,2,2,1) — the 288 GB-HBM memory-capability demonstration.
# ---------------------------------------------------------------------------

def mem_budget_gib(gshape, pshape, width, modes, out_t, num_blocks):
    """Rough fp32 peak-memory estimate per rank.  Forward saves per block
    ~2 full-width activations (block input + pre-gelu z; the native DFT
    autograd Functions save only dims), the projection head saves its input
    plus the 128-channel mid activation, and backward adds ~60% transient
    headroom; parameters + grads + Adam moments ride on top."""
    import numpy as _np
    vol_local = _np.prod([g // p for g, p in zip(gshape[2:5], pshape[2:5])]) * out_t
    act = width * vol_local * 4               # one full-width activation, bytes
    proj = 128 * vol_local * 4                # projection mid activation
    weights = num_blocks * (2 ** 3) * width * width * _np.prod(
        [m for m in modes[:-1]]) * modes[-1] * 8 / max(_np.prod(pshape[2:5]), 1)
    total = (1.6 * (num_blocks * 2.2 * act + act + proj)  # fwd saved + bwd live
             + 4 * weights)                               # w + grad + adam m,v
    return total / 2**30


def create_capability_script():
    g = 256
    width, modes, out_t, blocks = 20, (16, 16, 16, 8), 32, 4
    pshape = (1, 1, 2, 2, 2, 1)
    est = mem_budget_gib([1, 2, g, g, g, 1], pshape, width, modes, out_t, blocks)
    assert est < 288 * 0.85, f"256^3 config estimated {est:.0f} GiB/rank > HBM budget"
    fname = Path("capability_256cubed_gpu.sh")
    with open(fname, "w") as f:
        f.write(
            "#!/bin/bash\n"
            f"# 256^3 8-GPU model-parallel capability point (BASELINE config #4)\n"
            f"# estimated peak memory: ~{est:.0f} GiB/rank of 288 GiB HBM3E\n"
            "set -x\n"
            "python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 "
            "--master-addr 127.0.0.1 --master-port 29519 "
            f"../benchmarks/bench.py --input-shape 1 2 {g} {g} {g} 1 "
            f"--modes {' '.join(map(str, modes))} "
            f"--partition_shape {' '.join(map(str, pshape))} "
            f"--width {width} --num-timesteps {out_t} --device cuda "
            "--benchmark-type grad --output-dir capability_256cubed\n")
    os.chmod(fname, 0o755)
    print(f"created script: {fname.name} (est. {est:.0f} GiB/rank)")


create_capability_script()
