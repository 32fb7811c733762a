"""Randomized stress of the free-running interval engine (DP-3): varied
world sizes, fire periods and a pathologically slow last rank exercise the
catch-up posting, ring backpressure and the two-phase shutdown agreement.
Invariant: every rank posts AND applies the identical generation sequence
and ends with the identical aggregated gradient."""

import os
import random
import sys
import time

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from distributedmnist_amd.parallel.sync import SyncEngine  # noqa: E402

from mp_utils import init_pg, spawn_collect  # noqa: E402


def _run(rank, world, port, interval_ms, steps, seed, q):
    init_pg(rank, world, port)
    rng = random.Random(seed * 100 + rank)
    g = torch.ones(256)
    eng = SyncEngine(g, mode="interval", interval_ms=interval_ms,
                     rank=rank, world_size=world)
    for s in range(steps):
        eng.step_begin(s)
        if rank == world - 1 and rng.random() < 0.3:
            time.sleep(rng.uniform(0.02, 0.08))  # ring-pressure straggler
        else:
            time.sleep(rng.uniform(0, 0.004))
        eng.reduce(s, 0.001)
    for _ in eng.finalize_interval():
        pass
    q.put((rank, eng.generation, eng._gen_posted,
           float(eng.flat_grad.sum())))
    dist.destroy_process_group()


def _trial(world, interval_ms, steps, seed):
    res = spawn_collect(_run, world, args=(interval_ms, steps, seed))
    res = {r: tuple(v) for r, v in res.items()}
    assert len(set(res.values())) == 1, res
    return res[0]


@pytest.mark.timeout(300)
@pytest.mark.parametrize("world,interval_ms,steps", [
    (2, 5, 40),    # fast firing: catch-up + ring backpressure
    (3, 10, 25),   # 3 ranks, jittered
    (4, 5, 30),    # 4 ranks, heavy pressure
])
def test_interval_stress(world, interval_ms, steps):
    gen, posted, _ = _trial(world, interval_ms, steps, seed=world)
    assert gen == posted > 0


def _run_staggered(rank, world, port, q):
    # recreate the shutdown-deadlock geometry: rank 1 starts LATE (spawn
    # stagger), rank 0 finishes its few steps immediately and sits in
    # finalize phase 1 firing every 10 ms until its ring fills
    time.sleep(rank * 2.0)
    init_pg(rank, world, port)
    g = torch.ones(128)
    eng = SyncEngine(g, mode="interval", interval_ms=10.0,
                     rank=rank, world_size=world)
    for s in range(4):
        eng.step_begin(s)
        time.sleep(0.01)
        eng.reduce(s, 0.001)
    for _ in eng.finalize_interval():
        pass
    q.put((rank, eng.generation, eng._gen_posted))
    dist.destroy_process_group()


@pytest.mark.timeout(240)
def test_interval_shutdown_staggered_no_deadlock():
    """Regression: a rank ahead of the fire schedule used to hard-block in
    ring backpressure during finalize phase 1 (waiting on generations its
    stopped peer would never post) while the peer blocked in the phase-2
    MAX; a second shape had the phase-1 rank yielding forever because
    completed ring slots were never recycled.  Both fixed in sync.py."""
    res = _trial_raw(_run_staggered, 2)
    assert res[0] == res[1]


def _trial_raw(fn, world):
    res = spawn_collect(fn, world, collect_s=200)
    return {r: tuple(v) for r, v in res.items()}
