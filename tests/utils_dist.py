"""Multi-process test harness: spawn N ranks with gloo (CPU) or nccl (GPU).

Stands in for the reference's mpirun-launched smoke tests
(common/comm_core/tests/test_comm.py) — real pytest assertions, world_size=2
over gloo runs with no GPU (SURVEY.md §4 implication).
"""
import os
import pickle
import tempfile
import traceback

import torch.multiprocessing as mp


def _entry(rank, world, port, fn, args, resdir):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    try:
        out = fn(rank, world, *args)
        with open(os.path.join(resdir, f"rank{rank}.pkl"), "wb") as f:
            pickle.dump(("ok", out), f)
    except Exception as e:  # noqa: BLE001
        with open(os.path.join(resdir, f"rank{rank}.pkl"), "wb") as f:
            pickle.dump(("err", f"{e}\n{traceback.format_exc()}"), f)
        raise


def run_dist(fn, world_size=2, args=(), timeout=240):
    """Run fn(rank, world_size, *args) in world_size processes; return list of
    per-rank return values. Raises on any rank failure."""
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    with tempfile.TemporaryDirectory() as resdir:
        ctx = mp.start_processes(
            _entry, args=(world_size, port, fn, args, resdir),
            nprocs=world_size, join=True, start_method="spawn")
        outs = []
        for r in range(world_size):
            with open(os.path.join(resdir, f"rank{r}.pkl"), "rb") as f:
                status, val = pickle.load(f)
            if status != "ok":
                raise RuntimeError(f"rank {r} failed:\n{val}")
            outs.append(val)
        return outs
