"""Multi-process (fork + gloo) test harness: run fn(rank, world, *args) in
`world` processes on CPU and collect per-rank results."""
from __future__ import annotations

import multiprocessing as pymp
import os
import socket
import traceback


def free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _entry(rank, world, port, fn, fargs, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ.pop("RANK", None)
    os.environ.pop("WORLD_SIZE", None)
    import torch.distributed as dist
    try:
        res = fn(rank, world, *fargs)
        q.put((rank, res, None))
    except Exception:
        q.put((rank, None, traceback.format_exc()))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def run_dist(world: int, fn, fargs=(), timeout: float = 300.0) -> list:
    """Returns [result_rank0, result_rank1, ...]; raises on any failure.

    Uses the spawn start method: fork is unsafe once the pytest parent has
    run torch CPU ops (OpenMP pools) or initialized gloo — forked children
    inherit locked thread state and deadlock."""
    ctx = pymp.get_context("spawn")
    q = ctx.Queue()
    port = free_port()
    ps = [ctx.Process(target=_entry, args=(r, world, port, fn, fargs, q))
          for r in range(world)]
    for p in ps:
        p.start()
    results: dict[int, object] = {}
    errors = []
    for _ in range(world):
        try:
            rank, res, err = q.get(timeout=timeout)
        except Exception:
            for p in ps:
                p.terminate()
            raise TimeoutError(f"distributed test timed out; got {len(results)}"
                               f"/{world} results")
        if err:
            errors.append(f"rank {rank}:\n{err}")
        results[rank] = res
    for p in ps:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
    if errors:
        raise RuntimeError("\n".join(errors))
    return [results[r] for r in range(world)]
