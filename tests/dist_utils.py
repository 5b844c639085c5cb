"""Multi-process test harness: spawn world_size ranks over 127.0.0.1 TCP."""

from __future__ import annotations

import os
import socket

import torch.distributed as dist
import torch.multiprocessing as mp


def free_port() -> int:
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _worker(rank, world_size, port, fn, args, backend):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch_cgx_amd  # noqa: F401  (registers "cgx")
    dist.init_process_group(backend, rank=rank, world_size=world_size,
                            init_method=f"tcp://127.0.0.1:{port}")
    try:
        fn(rank, world_size, *args)
    finally:
        dist.destroy_process_group()


def run_dist(fn, world_size=2, backend="cgx", args=()):
    """Run fn(rank, world_size, *args) on world_size processes."""
    # spawn, not fork: earlier tests warm torch's OpenMP pool in the pytest
    # parent, and forking a threaded parent deadlocks DDP workers.
    start = os.environ.get("CGX_TEST_START_METHOD", "spawn")
    last = None
    for attempt in range(2):  # one retry shields against TCP port races
        port = free_port()
        try:
            mp.start_processes(_worker,
                               args=(world_size, port, fn, args, backend),
                               nprocs=world_size, join=True,
                               start_method=start)
            return
        except Exception as e:  # pragma: no cover - flake path
            last = e
            msg = str(e)
            transient = any(k in msg for k in
                            ("Address already in use", "bind", "ECONNREFUSED",
                             "Connection reset", "timed out"))
            if not transient or attempt == 1:
                raise
    raise last
