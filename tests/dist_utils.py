"""Helpers for multi-process (gloo, 127.0.0.1) distributed tests."""
import os
import socket

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def free_port() -> int:
    s = socket.socket()
    s.bind(('127.0.0.1', 0))
    port = s.getsockname()[1]
    s.close()
    return port


def setup_env(rank: int, world: int, port: int) -> None:
    os.environ['RANK'] = str(rank)
    os.environ['LOCAL_RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)


def run_dist(fn, world: int, timeout: float = 180.0, args=()):
    """Spawn `world` processes running fn(rank, world, port, *args)."""
    port = free_port()
    ctx = mp.get_context('spawn')
    procs = []
    q = ctx.Queue()
    for r in range(world):
        p = ctx.Process(target=_entry, args=(fn, r, world, port, q, args))
        p.start()
        procs.append(p)
    results = {}
    try:
        # drain BEFORE join: a child blocks on q.put of large payloads until
        # the parent consumes them
        for _ in range(world):
            r, val = q.get(timeout=timeout)
            results[r] = val
        for p in procs:
            p.join(timeout)
        for p in procs:
            assert p.exitcode == 0, f"subprocess exited {p.exitcode}"
    finally:
        for p in procs:
            if p.is_alive():
                p.terminate()
    return results


def _entry(fn, rank, world, port, q, args):
    setup_env(rank, world, port)
    torch.manual_seed(0)
    out = fn(rank, world, port, *args)
    if isinstance(out, torch.Tensor):
        out = out.detach().cpu().numpy()   # pickle by value (no fd sharing)
    q.put((rank, out))
    if dist.is_initialized():
        dist.destroy_process_group()
