"""Torch-profiler breakdown of one flagship training step (dev tool).

Usage (single GPU): python benchmarks/profile_step.py [--steps 3]
Prints the top ops by device time and a phase breakdown.
"""

import argparse
import sys
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import dfno_amd as dfno
from dfno_amd.partition import compute_distribution_info


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--width", type=int, default=20)
    p.add_argument("--num-blocks", type=int, default=4)
    p.add_argument("--grid", type=int, default=64)
    p.add_argument("--out", type=str, default=None)
    p.add_argument("--shapes", action="store_true")
    args = p.parse_args()

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    gs = [1, 2, args.grid, args.grid, args.grid, 1]
    _, P_x, _ = dfno.create_standard_partitions((1, 1, 1, 1, 1, 1))
    model = dfno.DistributedFNONd(P_x, gs, 30, args.width, (12, 12, 12, 8),
                                  num_blocks=args.num_blocks, device=device,
                                  dtype=torch.float32)
    criterion = dfno.DistributedRelativeLpLoss(P_x)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)

    info_x = compute_distribution_info(P_x, gs)
    out_shape = [gs[0], 1, *gs[2:-1], 30]
    x = torch.rand(*info_x["shape"], device=device)
    y_true = torch.rand(*out_shape, device=device)

    def step():
        opt.zero_grad(set_to_none=True)
        y = model(x)
        loss = criterion(y, y_true)
        loss.backward()
        opt.step()

    for _ in range(2):
        step()
    torch.cuda.synchronize()

    from torch.profiler import profile, ProfilerActivity

    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=args.shapes) as prof:
        for _ in range(args.steps):
            step()
        torch.cuda.synchronize()

    if args.shapes:
        table = prof.key_averages(group_by_input_shape=True).table(
            sort_by="self_cuda_time_total", row_limit=60)
    else:
        table = prof.key_averages().table(sort_by="self_cuda_time_total", row_limit=40)
    print(table)
    if args.out:
        with open(args.out, "w") as f:
            f.write(table)


if __name__ == "__main__":
    main()
