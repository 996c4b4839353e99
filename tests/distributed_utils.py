"""Helpers for multi-process CPU (gloo) tests.

The reference tested its distributed paths the same way — gloo over localhost
with torch.multiprocessing.spawn (/root/reference/assert.py:13-25,174-194);
here it is wrapped for pytest and the checks are per-shard with tight
tolerances (the reference's endpoint-only 1e-2 checks missed a real dk/dv
corruption — SURVEY.md §2.5/§4).
"""

from __future__ import annotations

import os
import pickle
import tempfile
import traceback

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank: int, world_size: int, port: int, fn, args, result_dir: str):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    torch.manual_seed(0)
    try:
        result = fn(rank, world_size, *args)
        with open(os.path.join(result_dir, f"rank{rank}.ok"), "wb") as f:
            pickle.dump(result, f)
    except Exception:
        with open(os.path.join(result_dir, f"rank{rank}.err"), "w") as f:
            f.write(traceback.format_exc())
        raise
    finally:
        dist.destroy_process_group()


_next_port = [29511]


def run_distributed(world_size: int, fn, *args):
    """Spawn `world_size` gloo ranks running fn(rank, world, *args); returns list of results."""
    port = _next_port[0]
    _next_port[0] += 1
    with tempfile.TemporaryDirectory() as result_dir:
        mp.spawn(_worker, args=(world_size, port, fn, args, result_dir),
                 nprocs=world_size, join=True)
        results = []
        for rank in range(world_size):
            err = os.path.join(result_dir, f"rank{rank}.err")
            if os.path.exists(err):
                raise AssertionError(f"rank {rank} failed:\n{open(err).read()}")
            with open(os.path.join(result_dir, f"rank{rank}.ok"), "rb") as f:
                results.append(pickle.load(f))
    return results
