"""Serving-path check: eager eval vs hipGraph replay (dfno_amd.serve).

Measures single-request latency and requests/s for the NS 2D+time and the
flagship 3D+time configs on one MI355X.  The graph path replays the whole
eval forward as ONE hipGraph launch; its win over eager scales with how
launch-bound the config is (small per-GPU grids / sharded serving).

Run on a GPU box:  python benchmarks/serve_check.py
"""

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

import dfno_amd as dfno  # noqa: E402
from dfno_amd.serve import GraphedEval  # noqa: E402


def clock(fn, iters=50, warmup=8):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def run(name, in_shape, t_out, width, modes, blocks=4):
    _, P_x, _ = dfno.create_standard_partitions((1,) * len(in_shape))
    model = dfno.DistributedFNONd(P_x, in_shape, t_out, width, modes,
                                  num_blocks=blocks,
                                  device=torch.device("cuda")).eval()
    x = torch.rand(*in_shape, device="cuda")
    with torch.no_grad():
        dt_e = clock(lambda: model(x))
    ge = GraphedEval(model, x)
    dt_g = clock(lambda: ge(x))
    print(f"{name:34s} eager {dt_e*1e3:7.3f} ms ({1/dt_e:7.1f} req/s)   "
          f"graph {dt_g*1e3:7.3f} ms ({1/dt_g:7.1f} req/s)   "
          f"{dt_e/dt_g:4.2f}x")


def main():
    torch.manual_seed(0)
    print("single-request eval latency (batch as configured), 1 MI355X")
    run("NS 2D+time 64^2 t10->40 w20", [1, 1, 64, 64, 10], 40, 20, (4, 4, 4))
    run("NS batch 10 (training batch)", [10, 1, 64, 64, 10], 40, 20, (4, 4, 4))
    run("flagship 3D 64^3 t30 w20",
        [1, 2, 64, 64, 64, 1], 30, 20, (12, 12, 12, 8))


if __name__ == "__main__":
    main()
