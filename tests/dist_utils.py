"""Helpers to run a function across N processes with the gloo backend
(CPU; the same code path runs over RCCL on the MI355X node)."""

import os
import pickle
import tempfile

import torch
import torch.multiprocessing as mp


def _worker(rank, world, fn, args, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        result = fn(rank, world, *args)
        with open(os.path.join(out_dir, f"out_{rank}.pkl"), "wb") as f:
            pickle.dump(result, f)
    finally:
        dist.destroy_process_group()


def run_dist(world: int, fn, *args, port: int = None):
    """Run fn(rank, world, *args) in `world` processes; returns [result_rank0,
    ..., result_rankN-1]."""
    if port is None:
        import socket
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()
    with tempfile.TemporaryDirectory() as out_dir:
        ctx = mp.get_context("spawn")
        procs = []
        for r in range(world):
            p = ctx.Process(target=_worker, args=(r, world, fn, args, port, out_dir))
            p.start()
            procs.append(p)
        for p in procs:
            p.join(timeout=300)
        for r, p in enumerate(procs):
            assert p.exitcode == 0, f"rank {r} exited with {p.exitcode}"
        outs = []
        for r in range(world):
            with open(os.path.join(out_dir, f"out_{r}.pkl"), "rb") as f:
                outs.append(pickle.load(f))
        return outs
