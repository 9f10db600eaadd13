"""2D+time Navier-Stokes FNO experiment.

MI355X-native counterpart of /root/reference/training/navier_stokes/
experiment_navier_stokes.py with its stale-API bugs fixed:
``generate_batch_indices`` is implemented (dfno_amd.utils) and the broken
``dim`` reference is replaced by explicit slicing.  Root loads the .mat
dataset (or synthesizes data with --synthetic), normalizes, splits, and
scatters shards with DistributedTranspose; training uses Adam + the
distributed MSE loss; per-rank checkpoints, .mat dumps and optional GIF
animation via gather-to-root.

Launch: python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
    --master-addr 127.0.0.1 training/navier_stokes/experiment_navier_stokes.py \
    --synthetic
"""

import argparse
import os
import sys
import time
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

import dfno_amd as dfno
from dfno_amd.partition import zero_volume_tensor

parser = argparse.ArgumentParser()
parser.add_argument("--input", "-i", type=Path, default=None)
parser.add_argument("--synthetic", action="store_true",
                    help="use random data instead of a .mat file")
parser.add_argument("--partition-shape", "-ps", type=int, default=(1, 1, 2, 2, 1), nargs=5)
parser.add_argument("--num-data", "-nd", type=int, default=1000)
parser.add_argument("--sampling-rate", "-sr", type=int, default=1)
parser.add_argument("--in-timesteps", "-it", type=int, default=10)
parser.add_argument("--out-timesteps", "-ot", type=int, default=40)
parser.add_argument("--train-split", "-ts", type=float, default=0.8)
parser.add_argument("--width", "-w", type=int, default=20)
parser.add_argument("--modes", "-m", type=int, default=(4, 4, 4), nargs=3)
parser.add_argument("--num-blocks", "-nb", type=int, default=4)
parser.add_argument("--num-epochs", "-ne", type=int, default=500)
parser.add_argument("--batch-size", "-bs", type=int, default=10)
parser.add_argument("--checkpoint-interval", "-ci", type=int, default=25)
parser.add_argument("--grid", type=int, default=64)
parser.add_argument("--generate-visualization", "-gv", action="store_true")
args = parser.parse_args()

dfno.init_distributed()
P_world, P_x, P_0 = dfno.create_standard_partitions(args.partition_shape)
use_cuda, _, _, device, _ = dfno.get_env(P_x)

torch.manual_seed(P_x.rank + 123)
np.random.seed(P_x.rank + 123)

# the reference runs NS training under autograd anomaly detection
# (experiment_navier_stokes.py:54); kept for behavioral parity
torch.set_anomaly_enabled(True)

# root-generated run timestamp broadcast to all ranks (reference :50-52)
B = dfno.Broadcast(P_0, P_x)
ts = torch.tensor([float(int(time.time()))], device=device) if P_0.active else \
    zero_volume_tensor(device=device)
timestamp = int(B(ts).item())

stem = args.input.stem if args.input is not None else "synthetic"
out_dir = Path(f"data/{stem}_{timestamp}")
if P_0.active:
    os.makedirs(out_dir, exist_ok=True)
    print(f"created output directory: {out_dir.resolve()}")

g = args.grid // args.sampling_rate
T_in, T_out = args.in_timesteps, args.out_timesteps

if P_0.active:
    if args.synthetic or args.input is None:
        u = torch.rand(args.num_data, 1, args.grid, args.grid, T_in + T_out,
                       dtype=torch.float32)
    else:
        from mat73 import loadmat

        u = torch.tensor(loadmat(str(args.input))["u"],
                         dtype=torch.float32)[: args.num_data].unsqueeze(1)
    sr = args.sampling_rate
    x_sl = (slice(None), slice(None), slice(None, None, sr), slice(None, None, sr),
            slice(None, T_in))
    y_sl = (slice(None), slice(None), slice(None, None, sr), slice(None, None, sr),
            slice(T_in, T_in + T_out))
    data = {}
    x, data["mu_x"], data["std_x"] = dfno.unit_guassian_normalize(u[x_sl])
    y, data["mu_y"], data["std_y"] = dfno.unit_guassian_normalize(u[y_sl])
    split = int(args.train_split * args.num_data)
    data["x_train"], data["x_test"] = x[:split], x[split:]
    data["y_train"], data["y_test"] = y[:split], y[split:]
    for k, v in data.items():
        print(f"{k}.shape = {tuple(v.shape)}")
    n_train, n_test = split, args.num_data - split
else:
    data = {k: zero_volume_tensor() for k in
            ["mu_x", "std_x", "mu_y", "std_y", "x_train", "x_test", "y_train", "y_test"]}
    split = int(args.train_split * args.num_data)
    n_train, n_test = split, args.num_data - split

# scatter shards root -> P_x (reference :91-93)
shapes = {
    "x_train": [n_train, 1, g, g, T_in], "x_test": [n_test, 1, g, g, T_in],
    "y_train": [n_train, 1, g, g, T_out], "y_test": [n_test, 1, g, g, T_out],
    "mu_x": [1, 1, g, g, T_in], "std_x": [1, 1, g, g, T_in],
    "mu_y": [1, 1, g, g, T_out], "std_y": [1, 1, g, g, T_out],
}
tensors = {}
for k in sorted(shapes):
    S = dfno.DistributedTranspose(P_0, P_x, global_shape=shapes[k])
    tensors[k] = S(data[k]).to(device)
x_train, x_test = tensors["x_train"], tensors["x_test"]
y_train, y_test = tensors["y_train"], tensors["y_test"]
mu_y, std_y = tensors["mu_y"], tensors["std_y"]
del data

print(f"index = {tuple(P_x.index)}, x_train.shape = {tuple(x_train.shape)}, "
      f"y_train.shape = {tuple(y_train.shape)}")

network = dfno.DistributedFNONd(
    P_x, [args.batch_size, 1, g, g, T_in], T_out, args.width, tuple(args.modes),
    num_blocks=args.num_blocks, device=device, dtype=torch.float32)
criterion = dfno.DistributedMSELoss(P_x)
mse = dfno.DistributedMSELoss(P_x)
optimizer = torch.optim.Adam(network.parameters(), lr=1e-3, weight_decay=1e-4)

steps, train_accs, test_accs = [], [], []

for i in range(args.num_epochs):
    network.train()
    batch_indices = dfno.generate_batch_indices(P_x, x_train.shape[0] if x_train.dim() > 1 else n_train,
                                                args.batch_size, shuffle=True, seed=i)
    train_loss, n_train_batch = 0.0, 0
    for j, (a, b) in enumerate(batch_indices):
        optimizer.zero_grad(set_to_none=True)
        y_hat = network(x_train[a:b])
        y = dfno.unit_gaussian_denormalize(y_train[a:b], mu_y, std_y)
        y_hat = dfno.unit_gaussian_denormalize(y_hat, mu_y, std_y)
        loss = criterion(y_hat, y)
        if P_0.active:
            print(f"epoch = {i}, batch = {j}, loss = {loss.item()}")
            train_loss += loss.item()
            n_train_batch += 1
        loss.backward()
        optimizer.step()

    if P_0.active and n_train_batch:
        print(f"epoch = {i}, average train loss = {train_loss / n_train_batch}")
        steps.append(i)
        train_accs.append(train_loss / n_train_batch)

    network.eval()
    with torch.no_grad():
        test_loss, test_mse, n_test_batch = 0.0, 0.0, 0
        y_true, y_pred = [], []
        for a, b in dfno.generate_batch_indices(P_x, x_test.shape[0] if x_test.dim() > 1 else n_test,
                                                args.batch_size, shuffle=False):
            y_hat = network(x_test[a:b])
            y = dfno.unit_gaussian_denormalize(y_test[a:b], mu_y, std_y)
            y_hat = dfno.unit_gaussian_denormalize(y_hat, mu_y, std_y)
            test_loss += criterion(y_hat, y).item()
            test_mse += mse(y_hat, y).item()
            y_true.append(y)
            y_pred.append(y_hat)
            n_test_batch += 1

    if P_0.active and n_test_batch:
        print(f"average test loss = {test_loss / n_test_batch}")
        print(f"average test mse  = {test_mse / n_test_batch}")
        test_accs.append(test_loss / n_test_batch)

    j = i + 1
    if j % args.checkpoint_interval == 0:
        with torch.no_grad():
            model_path = out_dir / f"model_{j:04d}_{max(P_x.rank, 0):04d}.pt"
            torch.save(network.state_dict(), model_path)
            print(f"saved model: {model_path.resolve()}")

            y_true_t = torch.cat(tuple(y_true)) if y_true else zero_volume_tensor()
            y_pred_t = torch.cat(tuple(y_pred)) if y_pred else zero_volume_tensor()
            try:
                from scipy import io

                io.savemat(out_dir / f"mat_{j:04d}_{max(P_x.rank, 0):04d}.mat",
                           {"y_true": y_true_t.cpu().numpy(),
                            "y_pred": y_pred_t.cpu().numpy()})
            except Exception:
                pass

            if args.generate_visualization:
                gshape = [n_test, 1, g, g, T_out]
                G1 = dfno.DistributedTranspose(P_x, P_0, global_shape=gshape)
                G2 = dfno.DistributedTranspose(P_x, P_0, global_shape=gshape)
                y_true_f = G1(y_true_t).cpu().numpy()
                y_pred_f = G2(y_pred_t).cpu().numpy()
                if P_0.active:
                    import matplotlib

                    matplotlib.use("Agg")
                    import matplotlib.pyplot as plt
                    from matplotlib.animation import FuncAnimation

                    fig = plt.figure()
                    ax1, ax2 = fig.add_subplot(121), fig.add_subplot(122)
                    im1 = ax1.imshow(np.squeeze(y_true_f[0, :, :, :, 0]), animated=True)
                    im2 = ax2.imshow(np.squeeze(y_pred_f[0, :, :, :, 0]), animated=True)

                    def animate(k):
                        im1.set_data(np.squeeze(y_true_f[0, :, :, :, k]))
                        im2.set_data(np.squeeze(y_pred_f[0, :, :, :, k]))
                        return (im1, im2)

                    ax1.title.set_text(r"$y_{true}$")
                    ax2.title.set_text(r"$y_{pred}$")
                    anim = FuncAnimation(fig, animate, frames=T_out, repeat=True)
                    anim.save(out_dir / f"anim_{j:04d}.gif")

                    fig = plt.figure()
                    ax = fig.add_subplot(111)
                    ax.plot(steps, train_accs, label="Average Train Loss")
                    ax.plot(steps, test_accs, label="Average Test Loss")
                    plt.legend()
                    plt.xlabel("Epoch")
                    plt.ylabel("Loss")
                    plt.savefig(out_dir / f"curves_{j:04d}.png")

dfno.finalize_distributed()
