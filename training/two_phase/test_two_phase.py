"""Two-phase FNO inference + gather-to-root (reference:
/root/reference/training/two_phase/test_two_phase.py).

Loads the per-rank sharded checkpoints written by train_two_phase.py,
predicts on a sample, gathers x / y / y_hat to the root rank with
Repartition collectors, and writes plots + an h5 dump there.
"""

import argparse
import os
import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

import dfno_amd as dfno
from dfno_amd.data import DistributedSleipnerDataset3D
from dfno_amd.partition import compute_distribution_info


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--ranks", "-n", type=int, default=None)
    p.add_argument("--data", choices=["synthetic", "local", "azure"], default="synthetic")
    p.add_argument("--data-root", type=str, default=None)
    p.add_argument("--sample", type=int, default=1001)
    p.add_argument("--width", type=int, default=20)
    p.add_argument("--modes", type=int, nargs=4, default=(12, 12, 12, 8))
    p.add_argument("--shape", type=int, nargs=4, default=(60, 60, 64, 30))
    p.add_argument("--model-dir", type=str, default="data/")
    p.add_argument("--out-dir", type=str, default="data/")
    args = p.parse_args()

    dfno.init_distributed()
    n = args.ranks or max(1, int(os.environ.get("WORLD_SIZE", "1")))
    P_world, P_x, P_root = dfno.create_standard_partitions((1, 1, 1, n, 1, 1))
    _, _, _, device, _ = dfno.get_env(P_x)

    shape = tuple(args.shape)
    ds = DistributedSleipnerDataset3D(
        P_x, [args.sample], shape=shape, root=args.data_root,
        synthetic=args.data == "synthetic", normalize=True)
    x, y = ds[0]
    x = x.unsqueeze(0).to(device)
    y = y.unsqueeze(0).to(device)

    model = dfno.DistributedFNONd(P_x, [1, 2, *shape[:-1], 1], shape[-1],
                                  args.width, tuple(args.modes), device=device)
    ckpt = Path(args.model_dir) / f"model_{max(P_x.rank, 0):04d}.pt"
    if ckpt.exists():
        model.load_state_dict(torch.load(ckpt, map_location=device, weights_only=True))
        print(f"rank = {P_x.rank}, loaded {ckpt}")
    else:
        print(f"rank = {P_x.rank}, no checkpoint at {ckpt}; using random init")

    model.eval()
    with torch.no_grad():
        y_hat = model(x)

    # gather full tensors to root (reference test_two_phase.py:21-23,96-98)
    x_g = [1, 2, *shape[:-1], 1]
    y_g = [1, 1, *shape]
    collect_x = dfno.Repartition(P_x, P_root, global_shape=x_g)
    collect_y = dfno.Repartition(P_x, P_root, global_shape=y_g)
    collect_yh = dfno.Repartition(P_x, P_root, global_shape=y_g)
    xf = collect_x(x)
    yf = collect_y(y)
    yhf = collect_yh(y_hat)

    if P_root.active:
        out_dir = Path(args.out_dir)
        out_dir.mkdir(parents=True, exist_ok=True)
        xf, yf, yhf = xf.cpu(), yf.cpu(), yhf.cpu()
        try:
            import h5py

            with h5py.File(out_dir / "prediction.h5", "w") as f:
                f.create_dataset("x", data=xf.numpy())
                f.create_dataset("y", data=yf.numpy())
                f.create_dataset("y_hat", data=yhf.numpy())
        except ImportError:
            np.savez(out_dir / "prediction.npz", x=xf.numpy(), y=yf.numpy(),
                     y_hat=yhf.numpy())
        try:
            import matplotlib

            matplotlib.use("Agg")
            import matplotlib.pyplot as plt

            fig, axes = plt.subplots(1, 3, figsize=(14, 4))
            z = shape[2] // 2
            t = shape[3] - 1
            axes[0].imshow(xf[0, 0, :, :, z, 0])
            axes[0].set_title("permz")
            axes[1].imshow(yf[0, 0, :, :, z, t])
            axes[1].set_title("saturation (true)")
            axes[2].imshow(yhf[0, 0, :, :, z, t])
            axes[2].set_title("saturation (pred)")
            fig.savefig(out_dir / "prediction.png", dpi=120)
        except ImportError:
            pass
        err = (yhf - yf).norm() / yf.norm()
        print(f"relative L2 error: {float(err):.6f}")
        print(f"wrote {out_dir}/prediction.*")


if __name__ == "__main__":
    main()
    dfno.finalize_distributed()
