"""Multi-rank sharded scheduling — world_size 2 over gloo on CPU.

The distributed path the driver exercises on 8 GPUs (torchrun + RCCL) is
the same code; here two processes coordinate over gloo with
MASTER_ADDR=127.0.0.1.
"""

import json
import multiprocessing as mp
import os

import pytest
import torch

from volcano_amd.parallel.sharding import ShardingPolicy


def test_sharding_policy_partition():
    world = 4
    nodes = [f"n-{i:03d}" for i in range(100)]
    jobs = [f"default/job-{i}" for i in range(1000)]
    node_owners = [[] for _ in range(world)]
    for r in range(world):
        p = ShardingPolicy(r, world)
        node_owners[r] = p.filter_nodes(nodes)
    # disjoint and complete
    all_owned = sum(node_owners, [])
    assert sorted(all_owned) == sorted(nodes)
    assert len(set(all_owned)) == len(nodes)
    # jobs: disjoint & complete too
    counts = [0] * world
    for j in jobs:
        owners = [r for r in range(world)
                  if ShardingPolicy(r, world).owns_job(j)]
        assert len(owners) == 1
        counts[owners[0]] += 1
    # reasonably balanced (crc32 hash)
    assert min(counts) > 150


def _rank_main(rank, world, port, q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    try:
        from volcano_amd.parallel import DistributedScheduler, init_distributed
        from volcano_amd.scheduler import FakeBinder, SchedulerCache
        from volcano_amd.store import ObjectStore
        from volcano_amd.utils import synth

        policy = init_distributed(backend="gloo")
        GI = 1024 ** 3
        store = ObjectStore()
        for n in synth.make_nodes(8, cpu_milli=4000, mem=16 * GI):
            store.create("Node", n)
        store.create("Queue", synth.make_queue("default"))
        for j in range(10):
            synth.make_gang(store, f"g-{j}", replicas=2, cpu_milli=1000,
                            mem=GI)

        binder = FakeBinder()
        cache = SchedulerCache(store=store, binder=binder)
        ds = DistributedScheduler(cache, policy=policy)
        ds.run_once()

        stats = ds.last_stats
        my_nodes = set(cache.nodes)
        my_binds = dict(binder.binds)
        # binds only land on owned nodes
        assert all(n in my_nodes for n in my_binds.values())
        q.put((rank, int(stats[:, 0].sum()), sorted(my_nodes),
               sorted(my_binds)))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        import traceback
        q.put((rank, "ERROR", traceback.format_exc(), str(e)))


def test_two_rank_gloo_sharded_cycle():
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank_main, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = []
    for _ in range(2):
        results.append(q.get(timeout=120))
    for p in procs:
        p.join(timeout=30)
    errors = [r for r in results if r[1] == "ERROR"]
    assert not errors, errors
    results.sort()
    (_, total0, nodes0, binds0), (_, total1, nodes1, binds1) = results
    # both ranks agree on the global bound count via allgather
    assert total0 == total1 == 20          # 10 gangs × 2 pods all fit
    # shards are disjoint and complete
    assert not (set(nodes0) & set(nodes1))
    assert len(nodes0) + len(nodes1) == 8
    assert not (set(binds0) & set(binds1))
    assert len(binds0) + len(binds1) == 20


def test_four_rank_gloo_sharded_cycle():
    """World 4 — the driver's SCALE shape (minus GPUs): shards stay
    disjoint and the global count converges."""
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank_main, args=(r, 4, port, q))
             for r in range(4)]
    for p in procs:
        p.start()
    results = []
    for _ in range(4):
        results.append(q.get(timeout=180))
    for p in procs:
        p.join(timeout=30)
    errors = [r for r in results if r[1] == "ERROR"]
    assert not errors, errors
    results.sort()
    totals = {r[1] for r in results}
    assert totals == {20}                   # every rank agrees globally
    all_nodes = [n for r in results for n in r[2]]
    assert len(all_nodes) == len(set(all_nodes)) == 8
    all_binds = [b for r in results for b in r[3]]
    assert len(all_binds) == len(set(all_binds)) == 20
