"""Multi-GPU path logic covered on CPU: series sharding by series_id % N
 and the single per-bucket all-reduce, world_size=2 over
gloo.  Partials are computed with the oracle (allowed here: tests/), the
sharding + collective code mirrors bench.py's step()."""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from oracle import pyoracle as orc

NS = 1_000_000_000
T0 = 1_700_000_000_000_000_000


def _worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        nseries, npts = 8, 4096
        bucket_ns = 300 * NS
        nb = int(npts * NS // bucket_ns) + 1
        # shard: series_id % world == rank
        rng = np.random.default_rng(99)  # same stream on all ranks
        mx = torch.full((nb,), -np.inf, dtype=torch.float64)
        sm = torch.zeros(nb, dtype=torch.float64)
        ct = torch.zeros(nb, dtype=torch.int64)
        for s in range(nseries):
            ts = T0 + np.arange(npts, dtype=np.int64) * NS
            vals = np.round(np.clip(np.cumsum(rng.normal(0, 0.5, npts)) + 50, 0, 100), 1)
            if s % world != rank:
                continue
            m, su, c = orc.bucket_agg(ts, vals, None, T0, bucket_ns, nb)
            mx = torch.maximum(mx, torch.from_numpy(m))
            sm += torch.from_numpy(su)
            ct += torch.from_numpy(c)
        dist.all_reduce(sm)
        dist.all_reduce(ct)
        dist.all_reduce(mx, op=dist.ReduceOp.MAX)
        if rank == 0:
            q.put((mx.numpy(), sm.numpy(), ct.numpy()))
    finally:
        dist.destroy_process_group()


def test_sharded_bucket_allreduce_gloo():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29511
    procs = [ctx.Process(target=_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    got = q.get(timeout=240)
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    # unsharded truth
    nseries, npts = 8, 4096
    bucket_ns = 300 * NS
    nb = int(npts * NS // bucket_ns) + 1
    rng = np.random.default_rng(99)
    emx = np.full(nb, -np.inf)
    esm = np.zeros(nb)
    ect = np.zeros(nb, dtype=np.int64)
    for s in range(nseries):
        ts = T0 + np.arange(npts, dtype=np.int64) * NS
        vals = np.round(np.clip(np.cumsum(rng.normal(0, 0.5, npts)) + 50, 0, 100), 1)
        m, su, c = orc.bucket_agg(ts, vals, None, T0, bucket_ns, nb)
        emx = np.maximum(emx, m)
        esm += su
        ect += c
    gmx, gsm, gct = got
    assert (gct == ect).all()
    assert (gmx[ect > 0] == emx[ect > 0]).all()
    assert np.allclose(gsm, esm, rtol=1e-12)
