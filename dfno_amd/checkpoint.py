"""Offline checkpoint resharding: P-rank -> Q-rank without a process group.

The reference's per-rank sharded checkpoints (``model_{rank:04d}.pt``,
SURVEY.md section 5) can only be loaded by a job with the SAME partition
shape.  This tool re-cuts a checkpoint for a different partition: it
impersonates each source rank (``partition.as_rank``) to rebuild that
rank's exact module/shard geometry, assembles the global parameters
(root-stored linears, per-corner spectral weights via the blocks'
``corner_ids`` / ``corner_local_in_corner`` bookkeeping), and then
impersonates each destination rank to emit its shard files.

Pure CPU, no torch.distributed initialisation required (serial-mode
collectives are identities; only shard geometry depends on the rank).

CLI:
  python -m dfno_amd.checkpoint --src-dir D --src-partition 1 1 1 4 1 1 \
      --dst-partition 1 1 2 2 1 1 --shape 60 60 64 30 --width 20 \
      --modes 12 12 12 8 --channels-in 2 --out-dir D2
"""

from __future__ import annotations

import re
from pathlib import Path
from typing import Dict, List, Optional, Sequence

import torch

from .partition import Partition, as_rank

__all__ = ["reshard_checkpoint"]


def _build_model(pshape: Sequence[int], rank: int, in_shape, out_t: int,
                 width: int, modes, num_blocks: int):
    from .nn.fno import DistributedFNONd

    import numpy as np

    world = int(np.prod([int(s) for s in pshape]))
    with as_rank(rank):
        P_x = Partition(tuple(range(world)), pshape)
        model = DistributedFNONd(P_x, in_shape, out_t, width, modes,
                                 num_blocks=num_blocks, device="cpu",
                                 dtype=torch.float32)
    return model


def _shard_file(src_dir: Path, rank: int) -> Path:
    """Pick rank ``rank``'s newest shard file (final name preferred)."""
    final = src_dir / f"model_{rank:04d}.pt"
    if final.exists():
        return final
    cands = sorted(src_dir.glob(f"model_*_{rank:04d}.pt"))
    if not cands:
        raise FileNotFoundError(f"no model_*_{rank:04d}.pt under {src_dir}")
    return cands[-1]


def reshard_checkpoint(src_dir, src_pshape: Sequence[int],
                       dst_pshape: Sequence[int], in_shape, out_t: int,
                       width: int, modes, num_blocks: int, out_dir) -> List[Path]:
    """Re-cut a sharded checkpoint from ``src_pshape`` to ``dst_pshape``.

    Returns the list of written destination shard paths
    (``out_dir/model_{rank:04d}.pt``).
    """
    import numpy as np

    src_dir = Path(src_dir)
    out_dir = Path(out_dir)
    out_dir.mkdir(parents=True, exist_ok=True)
    src_world = int(np.prod([int(s) for s in src_pshape]))
    dst_world = int(np.prod([int(s) for s in dst_pshape]))

    # ---- pass 1: harvest global parameters from the source shards --------
    root_state: Dict[str, torch.Tensor] = {}   # root-stored (rank-0) params
    # corners[(block_idx, corner_id)] = full complex tensor [w, w, *corner]
    corners: Dict[tuple, torch.Tensor] = {}

    for r in range(src_world):
        model = _build_model(src_pshape, r, in_shape, out_t, width, modes,
                             num_blocks)
        state = torch.load(_shard_file(src_dir, r), map_location="cpu",
                           weights_only=False)
        model.load_state_dict(state)
        if r == 0:
            # root rank holds the full linear weights (and any other
            # non-spectral parameters/buffers worth carrying over verbatim)
            for k, v in state.items():
                if ".weights." not in k:
                    root_state[k] = v.clone()
        for bi, block in enumerate(model.blocks):
            for k, cid in enumerate(block.corner_ids):
                key = (bi, cid)
                if key not in corners:
                    corners[key] = torch.zeros(
                        width, width, *block.corner_shapes[k],
                        dtype=block.weights[k].dtype)
                sl = (slice(None), slice(None)) + tuple(
                    slice(a, b) for a, b in block.corner_local_in_corner[k])
                corners[key][sl] = block.weights[k].detach()

    # ---- pass 2: emit destination shards ---------------------------------
    written: List[Path] = []
    for q in range(dst_world):
        model = _build_model(dst_pshape, q, in_shape, out_t, width, modes,
                             num_blocks)
        state = model.state_dict()
        for k in state:
            if ".weights." not in k:
                if k not in root_state:
                    continue
                if q == 0:
                    # destination root takes the harvested full parameter
                    if tuple(root_state[k].shape) != tuple(state[k].shape):
                        raise RuntimeError(
                            f"{k}: shape {tuple(root_state[k].shape)} in "
                            f"source vs {tuple(state[k].shape)} expected")
                    state[k] = root_state[k]
                elif (state[k].numel() > 0
                      and tuple(state[k].shape) == tuple(root_state[k].shape)):
                    # replicated full-size state (batchnorm weight/bias/
                    # running stats, step counters) lives on EVERY rank, not
                    # just root — copy it so ranks stay synchronized; only
                    # zero-volume root-stored placeholders are kept local.
                    state[k] = root_state[k].clone()
        for bi, block in enumerate(model.blocks):
            for k, cid in enumerate(block.corner_ids):
                full = corners.get((bi, cid))
                if full is None:
                    raise RuntimeError(
                        f"block {bi} corner {cid} missing from source shards")
                sl = (slice(None), slice(None)) + tuple(
                    slice(a, b) for a, b in block.corner_local_in_corner[k])
                state[f"blocks.{bi}.weights.{k}"] = full[sl].clone()
        path = out_dir / f"model_{q:04d}.pt"
        torch.save(state, path)
        written.append(path)
    return written


def main(argv: Optional[List[str]] = None) -> None:
    import argparse

    p = argparse.ArgumentParser(description=__doc__.split("\n")[0])
    p.add_argument("--src-dir", required=True)
    p.add_argument("--out-dir", required=True)
    p.add_argument("--src-partition", type=int, nargs="+", required=True)
    p.add_argument("--dst-partition", type=int, nargs="+", required=True)
    p.add_argument("--shape", type=int, nargs="+", required=True,
                   help="global spatial+time shape, e.g. 60 60 64 30")
    p.add_argument("--channels-in", type=int, default=2)
    p.add_argument("--batch-size", type=int, default=1)
    p.add_argument("--width", type=int, default=20)
    p.add_argument("--modes", type=int, nargs="+", required=True)
    p.add_argument("--num-blocks", type=int, default=4)
    args = p.parse_args(argv)

    in_shape = [args.batch_size, args.channels_in, *args.shape[:-1], 1]
    out = reshard_checkpoint(args.src_dir, args.src_partition,
                             args.dst_partition, in_shape, args.shape[-1],
                             args.width, args.modes, args.num_blocks,
                             args.out_dir)
    for pth in out:
        print(f"wrote {pth}")


if __name__ == "__main__":
    main()
