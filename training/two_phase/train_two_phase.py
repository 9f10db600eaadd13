"""Two-phase CO2-flow FNO training (the flagship workload).

MI355X-native counterpart of /root/reference/training/two_phase/
train_two_phase.py: one process per GPU launched with torchrun
(RCCL over xGMI), spatial model parallelism over the Y axis by default.

  python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
      --master-addr 127.0.0.1 training/two_phase/train_two_phase.py \
      --data synthetic

Data backends: --data synthetic (default; random fields, no IO),
--data local --data-root <dir> (zarr directory store / h5 files), or
--data azure (reads CONTAINER/DATA_PATH/ACCOUNT_URL/SLEIPNER_CREDENTIALS
from the environment like the reference).

Checkpoints keep the reference's per-rank sharded layout and naming:
model_{epoch:04d}_{rank:04d}.pt every interval, model_{rank:04d}.pt at the
end, loss_epoch_<i>.h5 on root.
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
from dfno_amd.data import DistributedSleipnerDataset3D


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--ranks", "-n", type=int, default=None,
                   help="partition size along Y (default: world size)")
    p.add_argument("--data", choices=["synthetic", "local", "azure"], default="synthetic")
    p.add_argument("--data-root", type=str, default=None)
    p.add_argument("--num-train", type=int, default=800)
    p.add_argument("--num-valid", type=int, default=200)
    p.add_argument("--num-epochs", type=int, default=100)
    p.add_argument("--checkpoint-interval", type=int, default=10)
    p.add_argument("--resume", type=str, default=None, metavar="DIR",
                   help="resume from DIR's latest train_state_*.pt (model + "
                        "optimizer + epoch; per-rank sharded, same partition "
                        "shape required). The reference cannot resume "
                        "training (optimizer state never saved, SURVEY.md "
                        "section 5); this extends it.")
    p.add_argument("--batch-size", type=int, default=1)
    p.add_argument("--width", type=int, default=20)
    p.add_argument("--modes", type=int, nargs=4, default=(12, 12, 12, 8))
    p.add_argument("--shape", type=int, nargs=4, default=(60, 60, 64, 30),
                   help="global X Y Z T")
    p.add_argument("--lr", type=float, default=1e-3)
    p.add_argument("--heartbeat-timeout", type=float, default=0.0,
                   metavar="SEC", help="enable store-heartbeat rank-failure "
                   "detection: abort with RankFailure when a peer misses "
                   "heartbeats for SEC seconds (0 = off)")
    p.add_argument("--out-dir", type=str, default="data/")
    p.add_argument("--cache-dir", type=str, default=None)
    args = p.parse_args()

    dfno.init_distributed()
    n = args.ranks or max(1, int(os.environ.get("WORLD_SIZE", "1")))
    P_world, P_x, P_root = dfno.create_standard_partitions((1, 1, 1, n, 1, 1))
    use_cuda, _, _, device, _ = dfno.get_env(P_x)
    dtype = torch.float32

    torch.manual_seed(P_x.rank + 123)
    np.random.seed(P_x.rank + 123)

    shape = tuple(args.shape)
    nb = args.batch_size
    channel_in, channel_out = 2, 1

    client = None
    container = prefix = None
    root = args.data_root
    synthetic = args.data == "synthetic"
    if args.data == "azure":
        import azure.storage.blob  # optional dependency

        container = os.environ["CONTAINER"]
        prefix = os.environ["DATA_PATH"]
        client = azure.storage.blob.ContainerClient(
            account_url=os.environ["ACCOUNT_URL"],
            container_name=container,
            credential=os.environ["SLEIPNER_CREDENTIALS"])

    train_idx = list(range(1, args.num_train + 1))
    valid_idx = list(range(args.num_train + 1, args.num_train + args.num_valid + 1))
    mk = lambda idx: DistributedSleipnerDataset3D(
        P_x, idx, client=client, container=container, prefix=prefix,
        shape=shape, normalize=True, savepath=args.cache_dir,
        filename="sample", root=root, synthetic=synthetic)
    train_data, valid_data = mk(train_idx), mk(valid_idx)

    train_loader = torch.utils.data.DataLoader(train_data, batch_size=nb, shuffle=False)
    valid_loader = torch.utils.data.DataLoader(valid_data, batch_size=nb, shuffle=False)
    P_x.barrier()

    model = dfno.DistributedFNONd(
        P_x, [nb, channel_in, *shape[:-1], 1], shape[-1], args.width,
        tuple(args.modes), device=device, dtype=dtype)
    criterion = dfno.DistributedRelativeLpLoss(P_x)
    optimizer = torch.optim.Adam(model.parameters(), lr=args.lr)

    out_dir = Path(args.out_dir)
    if P_root.active:
        out_dir.mkdir(parents=True, exist_ok=True)
    train_accs, valid_accs = [], []

    start_epoch = 0
    if args.resume:
        rdir = Path(args.resume)
        rk = max(P_x.rank, 0)
        cands = sorted(rdir.glob(f"train_state_*_{rk:04d}.pt"))
        if not cands:
            raise FileNotFoundError(
                f"rank {rk}: no train_state_*_{rk:04d}.pt under {rdir}")
        state = torch.load(cands[-1], map_location=device, weights_only=False)
        model.load_state_dict(state["model"])
        optimizer.load_state_dict(state["optimizer"])
        start_epoch = int(state["epoch"])
        train_accs = list(state.get("train_accs", []))
        valid_accs = list(state.get("valid_accs", []))
        print(f"rank = {P_x.rank}, resumed epoch {start_epoch} from {cands[-1]}")
    P_x.barrier()

    hb = None
    if args.heartbeat_timeout > 0:
        from dfno_amd.health import HeartbeatMonitor
        hb = HeartbeatMonitor(interval=max(args.heartbeat_timeout / 8, 0.5),
                              timeout=args.heartbeat_timeout).start()

    for i in range(start_epoch, args.num_epochs):
        model.train()
        train_loss, n_train_batch = 0.0, 0
        for j, (x, y) in enumerate(train_loader):
            if hb is not None:
                hb.check()     # fail fast (RankFailure) on a dead peer
            optimizer.zero_grad(set_to_none=True)
            t0 = time.time()
            x = x.to(device)
            y = y.to(device)
            y_hat = model(x)
            loss = criterion(y_hat, y)
            if P_root.active:
                train_loss += loss.item()
                n_train_batch += 1
            loss.backward()
            optimizer.step()
            P_x.barrier()
            if use_cuda:
                torch.cuda.synchronize()
            print(f"epoch = {i}, batch = {j}, dt = {time.time() - t0}")

        if P_root.active and n_train_batch:
            train_accs.append(train_loss / n_train_batch)
        P_x.barrier()

        model.eval()
        valid_loss, n_valid_batch = 0.0, 0
        with torch.no_grad():
            for x, y in valid_loader:
                x = x.to(device)
                y = y.to(device)
                loss = criterion(model(x), y)
                if P_root.active:
                    valid_loss += loss.item()
                    n_valid_batch += 1

        if P_root.active and n_valid_batch:
            print(f"epoch = {i}, train loss = {train_accs[-1]:08f}, "
                  f"val loss = {valid_loss / n_valid_batch:08f}")
            valid_accs.append(valid_loss / n_valid_batch)

        if (i + 1) % args.checkpoint_interval == 0:
            if P_root.active:
                try:
                    import h5py

                    with h5py.File(out_dir / f"loss_epoch_{i}.h5", "w") as f:
                        f.create_dataset("train_loss", data=train_accs)
                        f.create_dataset("valid_loss", data=valid_accs)
                except ImportError:
                    np.savez(out_dir / f"loss_epoch_{i}.npz",
                             train_loss=train_accs, valid_loss=valid_accs)
            path = out_dir / f"model_{i + 1:04d}_{max(P_x.rank, 0):04d}.pt"
            torch.save(model.state_dict(), path)
            print(f"rank = {P_x.rank}, saved model: {path}")
            # full training state for --resume (model file above keeps the
            # reference's exact checkpoint layout; this adds optimizer+epoch)
            torch.save({"model": model.state_dict(),
                        "optimizer": optimizer.state_dict(),
                        "epoch": i + 1,
                        "train_accs": train_accs, "valid_accs": valid_accs},
                       out_dir / f"train_state_{i + 1:04d}_{max(P_x.rank, 0):04d}.pt")

    path = out_dir / f"model_{max(P_x.rank, 0):04d}.pt"
    torch.save(model.state_dict(), path)
    print(f"rank = {P_x.rank}, saved model after final iteration: {path}")
    if P_root.active:
        print("training finished.")
    if hb is not None:
        hb.stop()


if __name__ == "__main__":
    main()
    dfno.finalize_distributed()
