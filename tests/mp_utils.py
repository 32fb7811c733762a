"""Shared multiprocessing-test plumbing with HANG-PROOF collection.

Two flake classes this guards against (seen in CI-style repetition):
- gloo rendezvous stalls (default init timeout is 30 minutes): children
  pass an explicit short timeout to init_process_group;
- a child dying before q.put left the parent blocked forever in
  SimpleQueue.get: collect() polls with a timeout and fails fast when a
  child exits without reporting.
"""

import datetime
import os
import queue as pyqueue
import socket

import torch.distributed as dist
import torch.multiprocessing as mp


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def init_pg(rank, world, port, backend="gloo", timeout_s=120):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    dist.init_process_group(backend, rank=rank, world_size=world,
                            timeout=datetime.timedelta(seconds=timeout_s))


def spawn_collect(fn, world, args=(), join_s=120, collect_s=180):
    """Start `world` processes of fn(rank, world, port, *args, q); collect
    one q.put per rank; fail fast if a child dies silently."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = free_port()
    procs = [ctx.Process(target=fn, args=(r, world, port) + tuple(args) + (q,))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    import time
    deadline = time.time() + collect_s
    while len(results) < world:
        try:
            item = q.get(timeout=2)
        except pyqueue.Empty:
            dead = [i for i, p in enumerate(procs)
                    if not p.is_alive() and p.exitcode not in (0, None)]
            if dead:
                for p in procs:
                    if p.is_alive():
                        p.terminate()
                raise AssertionError(
                    f"child rank(s) {dead} died (exitcodes "
                    f"{[procs[i].exitcode for i in dead]}) before reporting")
            if time.time() > deadline:
                for p in procs:
                    if p.is_alive():
                        p.terminate()
                raise AssertionError("collection timed out (likely a "
                                     "rendezvous or collective hang)")
            continue
        r, *vals = item
        results[r] = vals
    for p in procs:
        p.join(join_s)
        assert p.exitcode == 0, p.exitcode
    return results
