"""Helpers to run multi-process gloo tests on CPU (world_size > 1 stands in
for a multi-GPU node, as the reference does with small MPI worlds)."""

import os
import tempfile
import traceback

import torch.multiprocessing as mp


def _worker(rank, world_size, fn, init_file, err_dir, args):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    try:
        dist.init_process_group(
            "gloo", init_method=f"file://{init_file}", rank=rank, world_size=world_size
        )
        fn(rank, world_size, *args)
    except Exception:
        with open(os.path.join(err_dir, f"rank{rank}.err"), "w") as f:
            f.write(traceback.format_exc())
        raise
    finally:
        if dist.is_initialized():
            try:
                dist.barrier()  # drain in-flight work before teardown (gloo race)
            except Exception:
                pass
            dist.destroy_process_group()


def run_dist(fn, world_size, *args, timeout=300):
    """Spawn ``world_size`` gloo processes each running fn(rank, world, *args).

    Retries once when the spawn dies without any rank writing a traceback
    (an infrastructure flake -- process startup under load -- rather than a
    test assertion, which would write rank<k>.err and re-fail anyway).
    """
    for attempt in range(2):
        with tempfile.TemporaryDirectory() as td:
            init_file = os.path.join(td, "init")
            try:
                mp.start_processes(
                    _worker,
                    args=(world_size, fn, init_file, td, args),
                    nprocs=world_size,
                    start_method="spawn",
                    join=True,
                )
                return
            except Exception:
                msgs = []
                for r in range(world_size):
                    p = os.path.join(td, f"rank{r}.err")
                    if os.path.exists(p):
                        msgs.append(f"--- rank {r} ---\n" + open(p).read())
                if not msgs and attempt == 0:
                    continue  # no rank-level traceback: infra flake, retry
                raise AssertionError(
                    "distributed test failed:\n" + "\n".join(msgs))
