"""Smoke demo: `python -m dfno_amd` (counterpart of the reference's
`python dfno/dfno.py` __main__ block, /root/reference/dfno/dfno.py:355-389).

Runs 10 timed fwd+bwd iterations of a 3D+time DistributedFNONd on synthetic
data.  Launch serially or under torchrun for a partitioned run, e.g.:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
      --master-addr 127.0.0.1 -m dfno_amd
"""

import os
import time

import torch

import dfno_amd as dfno
from dfno_amd.partition import compute_distribution_info


def main():
    dfno.init_distributed()
    ws = max(1, int(os.environ.get("WORLD_SIZE", "1")))
    shapes = {1: (1, 1, 1, 1, 1, 1), 2: (1, 1, 2, 1, 1, 1),
              4: (1, 1, 2, 2, 1, 1), 8: (1, 1, 4, 2, 1, 1)}
    P_world, P_x, P_root = dfno.create_standard_partitions(shapes.get(ws, (1, 1, ws, 1, 1, 1)))
    _, _, _, device, _ = dfno.get_env(P_x)

    width = 20
    modes = (4, 4, 4, 8)
    nt = 30
    in_shape = (1, 1, 64, 64, 64, 1)
    info = compute_distribution_info(P_x, in_shape)
    x = torch.rand(*info["shape"], device=device, dtype=torch.float32)

    network = dfno.DistributedFNONd(P_x, in_shape, nt, width, modes,
                                    num_blocks=4, device=device, dtype=x.dtype)
    criterion = dfno.DistributedMSELoss(P_x)
    y = network(x)

    for i in range(10):
        t0 = time.time()
        y = network(x)
        if device.type == "cuda":
            torch.cuda.synchronize()
        print(f"rank = {P_x.rank}, dt = {time.time() - t0}")

        loss = criterion(y, torch.rand_like(y))
        P_x.barrier()

        t0 = time.time()
        loss.backward()
        if device.type == "cuda":
            torch.cuda.synchronize()
        print(f"rank = {P_x.rank}, dt_grad = {time.time() - t0}")


if __name__ == "__main__":
    main()
