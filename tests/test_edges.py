"""Edge-case coverage: cron field semantics, store journal trimming,
watch kind filtering, resource vector edges, bit-plane overflow."""

import time

import pytest

from volcano_amd.api.resource import Resource, ResourceDims
from volcano_amd.scheduler.tensors import BitRegistry, NodeTensors
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth
from volcano_amd.utils.cron import CronSchedule


def test_cron_fields():
    # minute steps
    s = CronSchedule("*/15 * * * *")
    t = time.mktime((2026, 9, 12, 10, 30, 0, 0, 0, -1))
    assert s.matches(t)
    assert not s.matches(t + 60)
    # ranges + lists
    s = CronSchedule("0 9-17 * * 1-5")
    mon_10am = time.mktime((2026, 9, 14, 10, 0, 0, 0, 0, -1))   # Monday
    sun_10am = time.mktime((2026, 9, 13, 10, 0, 0, 0, 0, -1))   # Sunday
    assert s.matches(mon_10am)
    assert not s.matches(sun_10am)
    # next_after lands on a matching minute
    nxt = s.next_after(sun_10am)
    assert s.matches(nxt)
    with pytest.raises(ValueError):
        CronSchedule("* * *")


def test_store_journal_trim_and_filter():
    s = ObjectStore(journal_size=10)
    for i in range(25):
        s.create("Node", synth.make_node(f"n-{i:03d}"))
    evs = s.journal_since(0)
    assert len(evs) == 10                       # ring-trimmed
    assert evs[-1][0] == s.resource_version
    # kind filter
    s.create("Queue", synth.make_queue("q"))
    assert all(k == "Node" for _, _, k, _ in s.journal_since(0, ("Node",)))


def test_bit_registry_overflow():
    reg = BitRegistry(max_words=1)
    for i in range(64):
        reg.bit(f"k{i}")
    with pytest.raises(OverflowError):
        reg.bit("one-too-many")


def test_resource_from_vector_roundtrip_zero():
    dims = ResourceDims()
    r = Resource({})
    vec = r.to_vector(dims)
    assert Resource.from_vector(vec, dims).is_empty()


def test_node_tensors_dynamic_bit_width_growth():
    from volcano_amd.api.info import NodeInfo
    dims = ResourceDims()
    nt = NodeTensors(dims, label_words=2)
    nodes = [NodeInfo(n) for n in synth.make_nodes(4)]
    nt.pack(nodes)
    # burn through word 0 to force a second plane word
    for i in range(70):
        nt.add_dynamic_bit(f"grow-{i}", [0])
    assert nt.planes_t.shape[0] >= 2
    assert nt.planes_np.shape == tuple(nt.planes_t.shape)
    # require vector padding matches
    words = nt.bit_words([0, 69])
    assert len(words) == nt.labels.words


def test_agent_scheduler_skips_gang_pods():
    from volcano_amd.scheduler.agentscheduler import AgentScheduler
    store = ObjectStore()
    store.create("Node", synth.make_node("n1"))
    asched = AgentScheduler(store, workers=1)
    # a gang pod (has podgroup annotation) is NOT fast-path eligible
    store.create("Pod", synth.make_pod("gangpod", podgroup="pg1"))
    assert asched.run_once() == 0
