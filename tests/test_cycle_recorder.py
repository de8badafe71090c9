"""Per-cycle decision recorder (reference actions/allocate/recorder.go):
one JSON line per cycle to $VAMD_CYCLE_LOG."""

import json
import os


def test_cycle_recorder_writes_json_lines(tmp_path, monkeypatch):
    from volcano_amd.scheduler import FakeBinder, Scheduler, SchedulerCache, \
        default_config
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth

    log = tmp_path / "cycles.jsonl"
    monkeypatch.setenv("VAMD_CYCLE_LOG", str(log))
    store = ObjectStore()
    store.create("Queue", synth.make_queue("default"))
    for n in synth.make_nodes(4, cpu_milli=8000):
        store.create("Node", n)
    synth.make_gang(store, "rec", replicas=3, cpu_milli=1000)
    config = default_config()
    config.use_hip = False
    cache = SchedulerCache(store=store, binder=FakeBinder())
    sched = Scheduler(cache, config)
    sched.run_once()
    sched.run_once()

    lines = [json.loads(x) for x in log.read_text().splitlines()]
    assert len(lines) == 2
    first = lines[0]
    assert first["cycle"] == 1 and first["nodes"] == 4 and \
        first["jobs"] == 1 and first["pending_after"] == 0
    assert "allocate" in first["actions_ms"] and first["e2e_ms"] > 0
