"""Reference-protocol benchmark harness (dt / dt_comm / dt_comp / dt_grad).

MI355X-native counterpart of /root/reference/benchmarks/bench.py: same CLI,
same per-rank JSON output (one file named
``{input_shape}-{partition_shape}-{width}-{modes}-{nt}-{type}-{rank}-{size}.json``
with ``dt`` = one timed forward after a warm-up pass + barrier, ``dt_comm``
from the in-module timers, ``dt_comp = dt - dt_comm`` and, for
``--benchmark-type grad``, ``dt_grad`` = one timed backward), but launched
with torchrun (one rank per GPU over RCCL) instead of mpirun:

  python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
      --master-addr 127.0.0.1 benchmarks/bench.py \
      --input-shape 1 1 64 32 32 1 --partition_shape 1 1 2 2 1 1 \
      --modes 4 4 4 4 --num-timesteps 10 --device cuda --benchmark-type grad
"""

import argparse
import gc
import json
import os
import sys
import time
import traceback
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import dfno_amd as dfno
from dfno_amd.partition import compute_distribution_info


def dls(l, delimiter="_"):
    return delimiter.join(str(x) for x in l)


def print0(msg, P_0):
    if P_0.active:
        print(msg, flush=True)


def bench(input_shape, partition_shape, width, modes, nt, dev, ngpu,
          benchmark_type, output_dir=Path(".")):
    dfno.init_distributed()
    P_world, P_x, P_0 = dfno.create_standard_partitions(partition_shape)

    if dev == "cpu" or not torch.cuda.is_available():
        device = torch.device("cpu")
    else:
        _, _, _, device, _ = dfno.get_env(P_x)

    outfile = Path(f"{dls(input_shape)}-{dls(partition_shape)}-{width}-"
                   f"{dls(modes)}-{nt}-{benchmark_type}-{max(P_x.rank, 0)}-{P_x.size}.json")
    data = {}

    assert len(input_shape) == len(partition_shape)
    assert len(input_shape) - 2 == len(modes)
    assert width > 0 and nt > 0

    output_dir = Path(output_dir)
    if P_0.active and not output_dir.exists():
        output_dir.mkdir(parents=True, exist_ok=True)
        print(f"created output directory: {output_dir}")
    P_x.barrier()

    def sync():
        if device.type == "cuda":
            torch.cuda.synchronize()

    try:
        x_info = compute_distribution_info(P_x, input_shape)
        x = torch.rand(size=tuple(x_info["shape"]), device=device, dtype=torch.float32)
        network = dfno.DistributedFNONd(P_x, input_shape, nt, width, modes,
                                        device=device, dtype=torch.float32)
        network.eval()

        if benchmark_type == "eval":
            with torch.no_grad():
                print0("fake eval", P_0)
                y = network(x)
                sync()
                del y
                gc.collect()
                P_x.barrier()
                print0("real eval", P_0)
                t0 = time.time()
                y = network(x)
                sync()
                t1 = time.time()
                data["dt"] = t1 - t0
                data["dt_comm"] = network.dt_comm
                data["dt_comp"] = data["dt"] - data["dt_comm"]
        else:
            print0("fake eval+grad", P_0)
            y = network(x)
            y1 = torch.ones_like(y)
            y.backward(y1)
            sync()
            del y
            gc.collect()

            P_x.barrier()
            print0("real eval", P_0)
            t0 = time.time()
            y = network(x)
            sync()
            t1 = time.time()
            data["dt"] = t1 - t0
            data["dt_comm"] = network.dt_comm
            data["dt_comp"] = data["dt"] - data["dt_comm"]

            P_x.barrier()
            print0("real grad", P_0)
            t0 = time.time()
            y.backward(y1)
            sync()
            t1 = time.time()
            data["dt_grad"] = t1 - t0

        with open(output_dir / outfile, "w") as f:
            json.dump(data, f)
        print0(f"dt={data.get('dt'):.6f} dt_comm={data.get('dt_comm'):.6f} "
               f"dt_grad={data.get('dt_grad', float('nan')):.6f}", P_0)
    except Exception:
        traceback.print_exc()
        # one dead rank must not hang the job (reference bench.py:134-143)
        if dfno.is_distributed():
            import torch.distributed as dist

            dist.destroy_process_group()
        os._exit(1)


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--input-shape", "-is", type=int, nargs="+", required=True)
    parser.add_argument("--partition_shape", "-ps", type=int, nargs="+", required=True)
    parser.add_argument("--width", "-w", type=int, default=20)
    parser.add_argument("--modes", "-m", type=int, nargs="+", required=True)
    parser.add_argument("--num-timesteps", "-nt", type=int, default=10)
    parser.add_argument("--device", "-d", type=str, default="cuda")
    parser.add_argument("--num-gpus", "-ngpu", type=int, default=0)
    parser.add_argument("--benchmark-type", "-bt", type=str, default="eval",
                        choices=["eval", "grad"])
    parser.add_argument("--output-dir", "-o", type=Path, default=Path("."))
    args = parser.parse_args()

    bench(args.input_shape, args.partition_shape, args.width, args.modes,
          args.num_timesteps, args.device, args.num_gpus, args.benchmark_type,
          args.output_dir)
    dfno.finalize_distributed()
