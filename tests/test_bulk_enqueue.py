"""Bulk enqueue admission fast path == per-job votes.

The enqueue action may admit a queue's whole pending batch in one vote
when every enqueueable plugin's bulk counterpart agrees the total fits
(actions/enqueue.py).  Exactness argument: admission bounds are monotone
accumulated sums, so `total fits` <=> `every sequential prefix fits`.
These tests pin that equivalence (admission sets identical with the bulk
path disabled) and the fallback behavior when the batch does NOT fit.
"""

import random

from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def _admitted(store):
    return sorted(pg.meta.name for pg in store.list("PodGroup")
                  if pg.status.phase in ("Inqueue", "Running"))


def _run(bulk_enabled: bool, n_jobs: int, cpu_each: int, seed: int = 7):
    store = ObjectStore()
    cache = SchedulerCache(store=store, binder=FakeBinder())
    config = default_config()
    config.actions = ["enqueue"]
    sched = Scheduler(cache, config)
    if not bulk_enabled:
        orig_open = sched.open_session

        def patched():
            ssn = orig_open()
            ssn.job_enqueueable_bulk_fns.clear()
            return ssn
        sched.open_session = patched

    rng = random.Random(seed)
    for i in range(8):
        store.create("Node", synth.make_node(
            f"n-{i:02d}", cpu_milli=8000, mem=32 * GI))
    store.create("Queue", synth.make_queue("qa", weight=1))
    for j in range(n_jobs):
        synth.make_gang(store, f"g{j:03d}", replicas=rng.randint(1, 3),
                        queue="qa", cpu_milli=cpu_each, mem=GI,
                        phase="Pending")
    sched.run_once()
    return _admitted(store)


def test_bulk_path_matches_per_job_when_all_fit():
    # total demand well under cluster capacity -> bulk admits everything,
    # identically to the per-job vote loop
    a = _run(True, n_jobs=20, cpu_each=200)
    b = _run(False, n_jobs=20, cpu_each=200)
    assert a == b and len(a) == 20


def test_bulk_path_matches_per_job_when_batch_overflows():
    # total demand exceeds the overcommit bound -> bulk declines, action
    # falls back to per-job votes; the admitted prefix must be identical
    a = _run(True, n_jobs=40, cpu_each=4000)
    b = _run(False, n_jobs=40, cpu_each=4000)
    assert a == b
    assert 0 < len(a) < 40     # partial admission actually exercised


def test_bulk_accounting_persists_across_cycles():
    # jobs admitted via the bulk path must count against the next cycle's
    # inqueue accounting (phase carried through the store)
    store = ObjectStore()
    cache = SchedulerCache(store=store, binder=FakeBinder())
    config = default_config()
    config.actions = ["enqueue"]
    sched = Scheduler(cache, config)
    for i in range(4):
        store.create("Node", synth.make_node(
            f"n-{i}", cpu_milli=8000, mem=32 * GI))
    store.create("Queue", synth.make_queue("qa", weight=1))
    # cycle 1: batch fits exactly under overcommit (32000*1.2=38400 milli)
    for j in range(12):
        synth.make_gang(store, f"a{j:02d}", replicas=1, queue="qa",
                        cpu_milli=3000, mem=GI, phase="Pending")
    sched.run_once()
    assert len(_admitted(store)) == 12      # 36000 <= 38400
    # cycle 2: another 12 identical jobs; only floor(2400/3000)=0 more fit
    for j in range(12):
        synth.make_gang(store, f"b{j:02d}", replicas=1, queue="qa",
                        cpu_milli=3000, mem=GI, phase="Pending")
    sched.run_once()
    assert len(_admitted(store)) == 12      # headroom exhausted
