"""Multi-process (gloo, world_size>1) test harness.

Spawns worker processes that initialize the SPMD communicator over gloo on
127.0.0.1 and run a module-level function; results come back via a
multiprocessing queue. This is the CPU stand-in for the one-process-per-GPU
RCCL layout (the model for it: reference tests/test_ucx.py:35-107 runs the
comm context standalone inside a barrier job).
"""

from __future__ import annotations

import multiprocessing as mp
import os
import pickle
import random
import traceback
from typing import Any, Callable, List


def _worker(rank: int, world: int, port: int, fn_name: str, mod_name: str, args: tuple, q) -> None:
    try:
        os.environ["RANK"] = str(rank)
        os.environ["LOCAL_RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
        import importlib

        import sys
        sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

        from spark_rapids_ml_amd.parallel.context import init_comm, shutdown_comm

        init_comm(backend="gloo")
        mod = importlib.import_module(mod_name)
        fn = getattr(mod, fn_name)
        res = fn(*args)
        q.put((rank, "ok", pickle.dumps(res)))
        shutdown_comm()
    except Exception:
        q.put((rank, "err", traceback.format_exc()))


def run_distributed(fn: Callable, world_size: int = 2, args: tuple = ()) -> List[Any]:
    """Run module-level `fn(*args)` on `world_size` gloo ranks; returns the
    per-rank results ordered by rank. Raises on any rank failure."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    # keep below the Linux ephemeral range (32768+): an active outbound
    # connection can hold a port there and the gloo rendezvous bind fails
    port = random.randint(20000, 32000)
    procs = []
    for r in range(world_size):
        p = ctx.Process(
            target=_worker,
            args=(r, world_size, port, fn.__name__, fn.__module__, args, q),
        )
        p.start()
        procs.append(p)
    import queue as _queue
    import time as _time

    results: dict = {}
    errors = []
    got = 0
    deadline = _time.monotonic() + 300
    while got < world_size:
        try:
            rank, status, payload = q.get(timeout=10)
        except _queue.Empty:
            # a rank that died without reporting (crash/_exit) would block
            # the queue forever — detect it and fail instead of hanging
            dead = [r for r, p in enumerate(procs) if not p.is_alive()]
            missing = [r for r in dead if r not in results]
            if missing and got >= world_size - len(missing):
                errors.append(f"rank(s) {missing} exited without reporting")
                break
            if _time.monotonic() > deadline:
                errors.append("timed out waiting for rank results")
                break
            continue
        got += 1
        if status == "ok":
            results[rank] = pickle.loads(payload)
        else:
            errors.append(f"rank {rank}:\n{payload}")
    for p in procs:
        p.join(timeout=60)
        if p.is_alive():
            p.terminate()
    if errors:
        raise RuntimeError("distributed test failed:\n" + "\n".join(errors))
    return [results[r] for r in range(world_size)]
