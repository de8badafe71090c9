"""Table-driven scheduler tests through the uthelper harness (reference
pkg/scheduler/uthelper pattern: declare world → run actions → CheckAll)."""

import pytest

from volcano_amd.utils import synth
from volcano_amd.utils.uthelper import TestCommonStruct

GI = 1024 ** 3

CASES = [
    TestCommonStruct(
        name="gang-fits",
        nodes=synth.make_nodes(2, cpu_milli=2000, mem=8 * GI),
        queues=[synth.make_queue("default")],
        podgroups=[synth.make_podgroup("g1", min_member=3)],
        pods=[synth.make_pod(f"g1-w-{i}", "g1", cpu_milli=1000, mem=GI)
              for i in range(3)],
        expect_bind_count=3,
        expect_status={"default/g1": "Running"},
    ),
    TestCommonStruct(
        name="gang-too-big-reverts",
        nodes=synth.make_nodes(1, cpu_milli=2000, mem=8 * GI),
        queues=[synth.make_queue("default")],
        podgroups=[synth.make_podgroup("g2", min_member=3)],
        pods=[synth.make_pod(f"g2-w-{i}", "g2", cpu_milli=1000, mem=GI)
              for i in range(3)],
        expect_bind_count=0,
        expect_status={"default/g2": "Inqueue"},
    ),
    TestCommonStruct(
        name="selector-pins-node",
        nodes=[synth.make_node("a", labels={"zone": "z1"}),
               synth.make_node("b", labels={"zone": "z2"})],
        queues=[synth.make_queue("default")],
        podgroups=[synth.make_podgroup("g3", min_member=1)],
        pods=[synth.make_pod("g3-w-0", "g3", cpu_milli=500, mem=GI,
                             node_selector={"zone": "z2"})],
        expect_bind_map={"default/g3-w-0": "b"},
    ),
    TestCommonStruct(
        name="capacity-tier",
        tiers=[["priority", "gang", "conformance"],
               ["overcommit", "drf", "predicates", "capacity", "nodeorder",
                "binpack"]],
        nodes=synth.make_nodes(1, cpu_milli=4000, mem=32 * GI),
        queues=[synth.make_queue("default")],
        podgroups=[synth.make_podgroup("g4", min_member=1)],
        pods=[synth.make_pod(f"g4-w-{i}", "g4", cpu_milli=1000, mem=GI)
              for i in range(6)],
        expect_bind_count=4,      # elastic up to cluster capacity
    ),
]


@pytest.mark.parametrize("case", CASES, ids=lambda c: c.name)
def test_uthelper_cases(case):
    case.run().check_all()
