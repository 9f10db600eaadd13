"""Distributed Sleipner CO2-flow dataset (two-phase 3D training data).

Counterpart of /root/reference/training/two_phase/sleipner_dataset.py: each
rank reads only its Y-slab of every sample (the partition's dim-3 block),
assembles x = (permz, tops-repeated-over-z) as [2, X, Yloc, Z, 1] and
y = saturation as [1, X, Yloc, Z, T], and normalizes with GLOBAL min/max
computed via an allreduce over the partition (RCCL/gloo instead of the
reference's raw MPI allreduce, sleipner_dataset.py:93-96).

Backends:
* ``root`` (local): a zarr directory store or directory of .h5 files with
  arrays ``permz_<i>`` [X,Y,Z], ``tops_<i>`` [X,Y], ``saturation_<i>``
  [T,X,Y,Z] (the reference's Azure layout, minus the network).
* ``client`` (optional): any zarr-compatible store/mapping, e.g. an Azure
  container store, passed through unchanged.
* ``synthetic=True``: deterministic random fields per index (no IO), for
  benchmarks and tests.

Per-rank h5 caching mirrors the reference (``<filename>_<i>_<rank>.h5``).
"""

from __future__ import annotations

import os
from typing import Optional, Sequence

import numpy as np
import torch
from torch.utils.data import Dataset

from ..partition import Partition, _balanced_splits


class DistributedSleipnerDataset3D(Dataset):
    """Distributed dataset of two-phase flow samples, sharded along Y."""

    def __init__(self,
                 P_feat: Partition,
                 samples: Sequence[int],
                 client=None,
                 container: str = None,
                 prefix: str = None,
                 shape=(60, 60, 64, 30),
                 normalize: bool = True,
                 padding=None,
                 savepath: Optional[str] = None,
                 filename: Optional[str] = None,
                 keep_data: bool = False,
                 root: Optional[str] = None,
                 synthetic: bool = False):
        self.P_feat = P_feat
        self.samples = list(int(s) for s in samples)
        self.client = client
        self.container = container
        self.prefix = prefix
        self.shape = tuple(int(s) for s in shape)
        self.normalize = normalize
        self.padding = padding
        self.savepath = savepath
        self.filename = filename or "sample"
        self.keep_data = keep_data
        self.root = root
        self.synthetic = synthetic

        # this rank's Y-slab (partition dim 3 indexes Y in the reference's
        # [b, c, x, y, z, t] layout)
        ny = self.shape[1]
        py = int(P_feat.shape[3])
        yi = int(P_feat.index[3]) if P_feat.active else 0
        self.yStart, self.yEnd = _balanced_splits(ny, py)[yi]

        self.cache = None
        if savepath is not None:
            os.makedirs(savepath, exist_ok=True)
            self.cache = set()
            existing = set(os.listdir(savepath))
            for i in self.samples:
                f = self._cache_name(i)
                if f in existing:
                    self.cache.add(f)

        self._store = None
        if not synthetic and root is None and client is not None:
            import zarr

            try:  # modern zarr ABSStore-style path
                self._store = zarr.ABSStore(container=container, prefix=prefix,
                                            client=client)
            except Exception:
                self._store = client

    def _cache_name(self, i: int) -> str:
        return f"{self.filename}_{i:04d}_{max(self.P_feat.rank, 0):04d}.h5"

    def __len__(self):
        return len(self.samples)

    # -- raw field readers -------------------------------------------------
    def _read_fields(self, i: int):
        nx, ny, nz, nt = self.shape
        ys, ye = self.yStart, self.yEnd
        if self.synthetic:
            g = torch.Generator().manual_seed(i)
            permz = torch.rand(nx, ny, nz, generator=g)[:, ys:ye, :]
            tops = torch.rand(nx, ny, generator=g)[:, ys:ye]
            sat = torch.rand(nt + 1, nx, ny, nz, generator=g)[:, :, ys:ye, :]
            return permz, tops, sat
        if self.root is not None:
            return self._read_local(i)
        import zarr

        permz = torch.tensor(np.asarray(
            zarr.core.Array(self._store, path=f"permz_{i}")[:, ys:ye, :]),
            dtype=torch.float32)
        tops = torch.tensor(np.asarray(
            zarr.core.Array(self._store, path=f"tops_{i}")[:, ys:ye]),
            dtype=torch.float32)
        sat = torch.tensor(np.asarray(
            zarr.core.Array(self._store, path=f"saturation_{i}")[:self.shape[-1] + 1, :, ys:ye, :]),
            dtype=torch.float32)
        return permz, tops, sat

    def _read_local(self, i: int):
        ys, ye = self.yStart, self.yEnd
        h5path = os.path.join(self.root, f"sample_{i:04d}.h5")
        if os.path.exists(h5path):
            import h5py

            with h5py.File(h5path, "r") as f:
                permz = torch.tensor(f["permz"][:, ys:ye, :], dtype=torch.float32)
                tops = torch.tensor(f["tops"][:, ys:ye], dtype=torch.float32)
                sat = torch.tensor(f["saturation"][: self.shape[-1] + 1, :, ys:ye, :],
                                   dtype=torch.float32)
            return permz, tops, sat
        import zarr

        store = zarr.DirectoryStore(self.root)
        permz = torch.tensor(np.asarray(
            zarr.core.Array(store, path=f"permz_{i}")[:, ys:ye, :]), dtype=torch.float32)
        tops = torch.tensor(np.asarray(
            zarr.core.Array(store, path=f"tops_{i}")[:, ys:ye]), dtype=torch.float32)
        sat = torch.tensor(np.asarray(
            zarr.core.Array(store, path=f"saturation_{i}")[: self.shape[-1] + 1, :, ys:ye, :]),
            dtype=torch.float32)
        return permz, tops, sat

    def _global_minmax(self, t: torch.Tensor):
        mn = self.P_feat.allreduce_scalar(float(t.min()) if t.numel() else float("inf"), "min")
        mx = self.P_feat.allreduce_scalar(float(t.max()) if t.numel() else float("-inf"), "max")
        return mn, mx

    def __getitem__(self, index: int):
        i = self.samples[index]

        fname = self._cache_name(i)
        multi = self.P_feat.active and int(np.prod(self.P_feat.shape)) > 1
        if multi and torch.utils.data.get_worker_info() is not None:
            # forked DataLoader workers must not issue collectives: they
            # would race/desync the parent's process group
            raise RuntimeError(
                "DistributedSleipnerDataset3D requires num_workers=0 in a "
                "multi-rank partition (collectives run in __getitem__)")
        if self.cache is not None and multi:
            # make the cache-hit decision COLLECTIVE so a per-rank cache
            # divergence (partial prior run) cannot desync the normalization
            # allreduces below: all ranks take the same branch.
            hit = self.P_feat.allreduce_scalar(
                1.0 if fname in self.cache else 0.0, "min")
            if hit < 1.0:
                self.cache.discard(fname)
        if self.cache is not None and fname in self.cache:
            import h5py

            with h5py.File(os.path.join(self.savepath, fname), "r") as f:
                return (torch.tensor(np.asarray(f["x"])),
                        torch.tensor(np.asarray(f["y"])))

        permz, tops, sat = self._read_fields(i)
        sat = sat.permute(1, 2, 3, 0)[:, :, :, 1:]  # TXYZ -> XYZT, drop t=0
        nx, ny, nz, nt = sat.shape

        sat = sat.clamp_min(0)  # clip unphysical negatives
        if self.normalize:
            fields = {"permz": permz, "tops": tops, "sat": sat}
            for k, v in fields.items():
                mn, mx = self._global_minmax(v)
                v -= mn
                if mx - mn != 0:
                    v /= mx - mn
            permz, tops, sat = fields["permz"], fields["tops"], fields["sat"]

        # assemble channels: C x X x Y x Z x T layout (reference :99-111)
        permz_c = permz.reshape(1, nx, ny, nz, 1)
        tops_c = tops.reshape(1, nx, ny, 1, 1).repeat(1, 1, 1, nz, 1)
        x = torch.cat((permz_c, tops_c), dim=0)
        y = sat.reshape(1, nx, ny, nz, nt)

        if self.cache is not None:
            import h5py

            with h5py.File(os.path.join(self.savepath, fname), "w") as f:
                f.create_dataset("x", data=x.numpy())
                f.create_dataset("y", data=y.numpy())
            self.cache.add(fname)

        return x, y
