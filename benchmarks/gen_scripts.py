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
