"""HIP kernels vs the torch oracle — decision-exact on random inputs.

Every test builds random node/class state, runs the HIP kernel on
cuda:0 and reference.py on CPU, and requires identical *decisions*
(placements, caps, feasibility masks) and tight numeric agreement on
scores.
"""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

ref = pytest.importorskip("volcano_amd.ops.reference")


@pytest.fixture(scope="module")
def hip():
    from volcano_amd.ops import hip as h
    h._load()
    return h


def rand_state(N=10000, R=8, W=2, seed=0, device="cuda"):
    g = torch.Generator().manual_seed(seed)
    alloc = (torch.rand(N, R, generator=g) * 100 + 10)
    used = alloc * torch.rand(N, R, generator=g) * 0.9
    extra = torch.rand(N, R, generator=g) * 5
    ready = torch.rand(N, generator=g) > 0.05
    taints = (torch.rand(N, generator=g) * 8).to(torch.int64) & 0b11
    planes = (torch.rand(N, W, generator=g) * (2 ** 16)).to(torch.int64)
    req = torch.rand(R, generator=g) * 5
    req[torch.rand(R, generator=g) < 0.3] = 0.0
    return dict(alloc=alloc, used=used, extra=extra, ready=ready,
                taints=taints, planes=planes, req=req)


@pytest.mark.parametrize("seed", [0, 1, 2])
@pytest.mark.parametrize("N,R", [(177, 8), (10000, 8), (2000, 20),
                                 (1000, 61)])
def test_score_cap_matches_oracle(hip, seed, N, R):
    # R > 16 guards the regression where a fixed frac[16] register array
    # overflowed per-thread scratch once synthetic dims (paa:/hp:) grew
    # the registry past 16
    st = rand_state(N=N, R=R, seed=seed)
    R = st["req"].shape[0]
    W = st["planes"].shape[1]
    require = torch.tensor([0b1010, 0], dtype=torch.int64)
    forbid = torch.tensor([0b0100, 0], dtype=torch.int64)
    dim_w = torch.rand(R) + 0.5
    tol = 0b01

    score_c = torch.empty(N)
    cap_c = torch.empty(N, dtype=torch.int32)
    ref.score_cap(st["alloc"], st["used"], st["extra"], st["ready"],
                  st["taints"], st["planes"], st["req"], tol, require, forbid,
                  1.0, 0.7, 0.3, dim_w, None, score_c, cap_c)

    dev = "cuda"
    alloc_t = st["alloc"].t().contiguous().to(dev)
    used_t = st["used"].t().contiguous().to(dev)
    extra_t = st["extra"].t().contiguous().to(dev)
    planes_t = st["planes"].t().contiguous().to(dev)
    score_g = torch.empty(N, device=dev)
    cap_g = torch.empty(N, dtype=torch.int32, device=dev)
    hip.score_cap(alloc_t, used_t, extra_t,
                  st["ready"].to(torch.uint8).to(dev),
                  st["taints"].to(dev), planes_t, st["req"].to(dev), tol,
                  require.to(dev), forbid.to(dev), 1.0, 0.7, 0.3,
                  dim_w.to(dev), None, score_g, cap_g)
    torch.cuda.synchronize()

    score_g = score_g.cpu()
    cap_g = cap_g.cpu()
    feas_c = score_c > float("-inf")
    feas_g = score_g > float("-inf")
    assert torch.equal(feas_c, feas_g), "feasibility mask mismatch"
    assert torch.equal(cap_c, cap_g), "capacity mismatch"
    assert torch.allclose(score_c[feas_c], score_g[feas_g], atol=1e-4,
                          rtol=1e-4), "score mismatch"


@pytest.mark.parametrize("seed", [0, 1, 2, 3])
def test_select_commit_matches_oracle(hip, seed):
    N, R = 5000, 6
    g = torch.Generator().manual_seed(seed + 100)
    score = torch.rand(N, generator=g)
    score[torch.rand(N, generator=g) < 0.4] = float("-inf")
    cap = (torch.rand(N, generator=g) * 4).to(torch.int32) + 1
    cap[score == float("-inf")] = 0
    req = torch.rand(R, generator=g) * 3 + 0.5
    ntasks = 800
    qlimit = torch.full((R,), 1.0e18)
    qlimit[0] = float(ntasks) * req[0] * 0.6    # quota binds on dim 0
    K = min(ntasks, N)

    def run_cpu():
        used = torch.zeros(N, R)
        qa = torch.zeros(R)
        ln = torch.zeros(K, dtype=torch.int32)
        lc = torch.zeros(K, dtype=torch.int32)
        ll = torch.zeros((), dtype=torch.int32)
        pl = torch.zeros((), dtype=torch.int32)
        jp = torch.zeros((), dtype=torch.int32)
        ref.select_commit(score.clone(), cap, req, ntasks, used, qa, qlimit,
                          ln, lc, ll, pl, jp)
        return used, qa, ln, lc, ll, pl

    def run_gpu():
        dev = "cuda"
        used_t = torch.zeros(R, N, device=dev)
        qa = torch.zeros(R, device=dev)
        ln = torch.zeros(K, dtype=torch.int32, device=dev)
        lc = torch.zeros(K, dtype=torch.int32, device=dev)
        ll = torch.zeros(1, dtype=torch.int32, device=dev)
        pl = torch.zeros(1, dtype=torch.int32, device=dev)
        jp = torch.zeros(1, dtype=torch.int32, device=dev)
        hip.select_commit(score.clone().to(dev), cap.to(dev), req.to(dev),
                          ntasks, used_t, qa, qlimit.to(dev), ln, lc, ll, pl,
                          jp, -1)
        torch.cuda.synchronize()
        return (used_t.t().cpu(), qa.cpu(), ln.cpu(), lc.cpu(), ll.cpu(),
                pl.cpu())

    used_c, qa_c, ln_c, lc_c, ll_c, pl_c = run_cpu()
    used_g, qa_g, ln_g, lc_g, ll_g, pl_g = run_gpu()

    assert int(pl_c) == int(pl_g), "placed count mismatch"
    # same (node, count) multiset — order may differ only among ties;
    # scores are random floats so ties are measure-zero: require equality
    m_c = int(ll_c)
    m_g = int(ll_g)
    ent_c = sorted(zip(ln_c[:m_c].tolist(), lc_c[:m_c].tolist()))
    ent_g = sorted(zip(ln_g[:m_g].tolist(), lc_g[:m_g].tolist()))
    assert ent_c == ent_g, "placement entries mismatch"
    assert torch.allclose(used_c, used_g, atol=1e-3)
    assert torch.allclose(qa_c, qa_g, atol=1e-2)


@pytest.mark.gpu
@pytest.mark.parametrize("seed,N,ntasks", [(0, 5000, 4000), (1, 50000, 40000),
                                           (2, 937, 900), (3, 20000, 20000)])
def test_bulk_select_matches_oracle(hip, seed, N, ntasks):
    """Sort-based bulk select (mega-bundle path) vs the torch oracle —
    EXACT log order required (the bundle apply walk assigns slots to jobs
    in stream order)."""
    R = 5
    g = torch.Generator().manual_seed(seed + 500)
    score = torch.rand(N, generator=g)
    # force tie groups so the stable (score desc, index asc) order is
    # actually exercised, not just measure-zero random floats
    score = (score * 50).floor() / 50
    score[torch.rand(N, generator=g) < 0.3] = float("-inf")
    cap = (torch.rand(N, generator=g) * 6).to(torch.int32)
    cap[score == float("-inf")] = 0
    req = torch.rand(R, generator=g) * 2 + 0.5
    qlimit = torch.full((R,), 1.0e18)
    if seed % 2:
        qlimit[1] = float(ntasks) * req[1] * 0.5     # binding quota
    K = min(ntasks, N)

    used = torch.zeros(N, R)
    qa = torch.zeros(R)
    ln = torch.zeros(K, dtype=torch.int32)
    lc = torch.zeros(K, dtype=torch.int32)
    ll = torch.zeros((), dtype=torch.int32)
    pl = torch.zeros((), dtype=torch.int32)
    jp = torch.zeros((), dtype=torch.int32)
    ref.select_commit(score.clone(), cap, req, ntasks, used, qa, qlimit,
                      ln, lc, ll, pl, jp)

    dev = "cuda"
    used_t = torch.zeros(R, N, device=dev)
    qa_g = torch.zeros(R, device=dev)
    ln_g = torch.zeros(K, dtype=torch.int32, device=dev)
    lc_g = torch.zeros(K, dtype=torch.int32, device=dev)
    ll_g = torch.zeros(1, dtype=torch.int32, device=dev)
    pl_g = torch.zeros(1, dtype=torch.int32, device=dev)
    jp_g = torch.zeros(1, dtype=torch.int32, device=dev)
    scratch = torch.empty(4 * N, dtype=torch.int32, device=dev)
    hip.select_commit(score.clone().to(dev), cap.to(dev), req.to(dev),
                      ntasks, used_t, qa_g, qlimit.to(dev), ln_g, lc_g,
                      ll_g, pl_g, jp_g, -1, sort_scratch=scratch)
    torch.cuda.synchronize()

    m = int(ll)
    assert int(pl) == int(pl_g.cpu())
    assert m == int(ll_g.cpu())
    # EXACT order match (not multiset): stream order feeds bundle slots
    assert ln[:m].tolist() == ln_g[:m].cpu().tolist()
    assert lc[:m].tolist() == lc_g[:m].cpu().tolist()
    assert torch.allclose(used, used_t.t().cpu(), atol=1e-3)
    assert torch.allclose(qa, qa_g.cpu(), atol=1e-2)


def test_bulk_select_fused_gang_revert(hip):
    """Bulk path with fuse_min > achievable placements: in-kernel
    discard — usage and queue untouched, placed 0, log counts zeroed."""
    N, R = 2000, 4
    g = torch.Generator().manual_seed(9)
    score = torch.rand(N, generator=g)
    cap = torch.ones(N, dtype=torch.int32)          # 2000 instances max
    req = torch.ones(R)
    dev = "cuda"
    used_t = torch.zeros(R, N, device=dev)
    qa = torch.zeros(R, device=dev)
    qlimit = torch.full((R,), 1.0e18)
    K = N
    ln = torch.zeros(K, dtype=torch.int32, device=dev)
    lc = torch.zeros(K, dtype=torch.int32, device=dev)
    ll = torch.zeros(1, dtype=torch.int32, device=dev)
    pl = torch.zeros(1, dtype=torch.int32, device=dev)
    jp = torch.zeros(1, dtype=torch.int32, device=dev)
    scratch = torch.empty(4 * N, dtype=torch.int32, device=dev)
    hip.select_commit(score.to(dev), cap.to(dev), req.to(dev), 3000,
                      used_t, qa, qlimit.to(dev), ln, lc, ll, pl, jp,
                      2500, sort_scratch=scratch)   # needs 2500, only 2000
    torch.cuda.synchronize()
    assert int(pl.cpu()) == 0
    assert float(used_t.abs().sum().cpu()) == 0.0
    assert float(qa.abs().sum().cpu()) == 0.0
    assert int(lc.cpu().sum()) == 0                 # discard zeroed the log


def test_select_commit_gang_fused_revert(hip):
    """fuse_min above achievable → in-kernel revert leaves zero state."""
    N, R = 1000, 4
    g = torch.Generator().manual_seed(7)
    score = torch.rand(N, generator=g)
    cap = torch.ones(N, dtype=torch.int32)
    req = torch.ones(R)
    dev = "cuda"
    used_t = torch.zeros(R, N, device=dev)
    qa = torch.zeros(R, device=dev)
    K = 500
    ln = torch.zeros(K, dtype=torch.int32, device=dev)
    lc = torch.zeros(K, dtype=torch.int32, device=dev)
    ll = torch.zeros(1, dtype=torch.int32, device=dev)
    pl = torch.zeros(1, dtype=torch.int32, device=dev)
    jp = torch.zeros(1, dtype=torch.int32, device=dev)
    # 500 tasks but fuse_min 600 → must revert
    hip.select_commit(score.to(dev), cap.to(dev), req.to(dev), 500, used_t,
                      qa, torch.full((R,), 1e18, device=dev), ln, lc, ll, pl,
                      jp, 600)
    torch.cuda.synchronize()
    assert int(pl.cpu()) == 0
    assert float(used_t.abs().sum().cpu()) == 0.0
    assert float(qa.abs().sum().cpu()) == 0.0
    assert int(lc.abs().sum().cpu()) == 0


def test_cycle_runner_matches_torch_plan(hip):
    """Whole-plan equivalence: run_plan_hip vs run_plan_torch on a random
    synthetic inventory (multi-job, multi-class, queue quotas, gangs that
    must revert)."""
    from volcano_amd.scheduler import FakeBinder, Scheduler, SchedulerCache, \
        default_config
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth

    GI = 1024 ** 3

    def build(device, use_hip):
        store = ObjectStore()
        rng = np.random.RandomState(42)
        for i in range(500):
            store.create("Node", synth.make_node(
                f"n-{i:04d}", cpu_milli=float(rng.choice([8000, 16000, 32000])),
                mem=float(rng.choice([16, 32, 64])) * GI))
        store.create("Queue", synth.make_queue("qa", weight=2))
        store.create("Queue", synth.make_queue("qb", weight=1))
        rng2 = np.random.RandomState(7)
        for j in range(60):
            replicas = int(rng2.randint(1, 12))
            synth.make_gang(store, f"g-{j:03d}",
                            replicas=replicas,
                            min_member=max(1, replicas - int(rng2.randint(0, 3))),
                            queue="qa" if j % 2 == 0 else "qb",
                            cpu_milli=float(rng2.choice([500, 1000, 2000, 4000])),
                            mem=float(rng2.choice([1, 2, 4])) * GI)
        config = default_config()
        config.use_hip = use_hip
        config.device = device
        binder = FakeBinder()
        cache = SchedulerCache(store=store, binder=binder, device=device)
        Scheduler(cache, config).run_once()
        return binder.binds

    cpu_binds = build("cpu", False)
    gpu_binds = build("cuda", True)
    assert len(cpu_binds) > 0
    assert cpu_binds == gpu_binds, (
        f"CPU and GPU cycles disagree: {len(cpu_binds)} vs {len(gpu_binds)} binds")


def test_cycle_equivalence_feature_rich():
    """CPU vs GPU whole-cycle decisions on an inventory exercising taints,
    selectors, multi-role gangs, queues, priorities and anti-affinity."""
    from volcano_amd.api.objects import Taint, Toleration
    from volcano_amd.scheduler import FakeBinder, Scheduler, SchedulerCache, \
        default_config
    from volcano_amd.scheduler.config import PluginOption
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth

    GI = 1024 ** 3

    def build(device, use_hip):
        store = ObjectStore()
        rng = np.random.RandomState(99)
        for i in range(300):
            labels = {}
            taints = []
            if i % 3 == 0:
                labels["zone"] = f"z{i % 2}"
            if i % 17 == 0:
                taints.append(Taint("dedicated", "infra", "NoSchedule"))
            store.create("Node", synth.make_node(
                f"n-{i:04d}", cpu_milli=float(rng.choice([8000, 16000])),
                mem=float(rng.choice([16, 32])) * GI,
                labels=labels, taints=taints))
        store.create("Queue", synth.make_queue("qa", weight=3))
        store.create("Queue", synth.make_queue("qb", weight=1))
        rng2 = np.random.RandomState(5)
        for j in range(40):
            kind = j % 4
            q = "qa" if j % 2 == 0 else "qb"
            if kind == 0:       # plain gang
                synth.make_gang(store, f"g{j:03d}", replicas=int(rng2.randint(1, 9)),
                                queue=q, cpu_milli=1000, mem=GI,
                                priority=int(rng2.randint(0, 5)))
            elif kind == 1:     # selector gang
                synth.make_gang(store, f"g{j:03d}", replicas=3, queue=q,
                                cpu_milli=500, mem=GI,
                                node_selector={"zone": f"z{j % 2}"})
            elif kind == 2:     # tolerant gang with a shared host port
                synth.make_gang(store, f"g{j:03d}", replicas=2, queue=q,
                                cpu_milli=2000, mem=GI,
                                tolerations=[Toleration(
                                    key="dedicated", value="infra",
                                    effect="NoSchedule")])
                for pod in store.list("Pod"):
                    if pod.meta.name.startswith(f"g{j:03d}-"):
                        pod.host_ports = [7000 + (j % 3)]
                        store.update("Pod", pod)
            else:               # multi-role + anti-affinity
                pg = synth.make_podgroup(f"g{j:03d}", queue=q, min_member=3,
                                         min_task_member={"ps": 1,
                                                          "worker": 2})
                store.create("PodGroup", pg)
                ps = synth.make_pod(f"g{j:03d}-ps-0", f"g{j:03d}", queue=q,
                                    role="ps", cpu_milli=1000, mem=GI)
                ps.affinity = {"podAntiAffinity": {"group": f"aa{j}"}}
                store.create("Pod", ps)
                for w in range(2):
                    store.create("Pod", synth.make_pod(
                        f"g{j:03d}-w-{w}", f"g{j:03d}", queue=q,
                        role="worker", cpu_milli=500, mem=GI))
        config = default_config()
        config.use_hip = use_hip
        config.device = device
        config.tiers[1].plugins.append(PluginOption("interpodaffinity"))
        config.tiers[1].plugins.append(PluginOption(
            "task-topology", arguments={"affinity": [["ps", "worker"]]}))
        binder = FakeBinder()
        cache = SchedulerCache(store=store, binder=binder, device=device)
        sched = Scheduler(cache, config)
        sched.run_once()
        # late workers join the multi-role jobs: their cycle-2 classes get
        # task-topology bucket bias rows (per-class kernel bias path)
        for j in range(3, 40, 4):
            store.create("Pod", synth.make_pod(
                f"g{j:03d}-w-late", f"g{j:03d}",
                queue="qa" if j % 2 == 0 else "qb",
                role="worker", cpu_milli=500, mem=GI))
        sched.run_once()        # second cycle over mutated state
        return binder.binds

    cpu = build("cpu", False)
    gpu = build("cuda", True)
    assert len(cpu) > 50
    assert cpu == gpu, (
        f"decision divergence: {len(cpu)} vs {len(gpu)}; "
        f"diff={set(cpu.items()) ^ set(gpu.items())}")


def test_cycle_equivalence_subgroups(hip):
    """CPU vs GPU decisions for SubGroupPolicy jobs (per-subgroup gangs
    ride the fused in-kernel revert; minSubGroups gates post-run)."""
    from volcano_amd.scheduler import FakeBinder, Scheduler, SchedulerCache, \
        default_config
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth

    GI = 1024 ** 3

    def build(device, use_hip):
        store = ObjectStore()
        for i in range(40):
            store.create("Node", synth.make_node(
                f"n-{i:03d}", cpu_milli=4000, mem=16 * GI))
        store.create("Queue", synth.make_queue("default"))
        for j in range(6):
            pg = synth.make_podgroup(f"sg{j}", min_member=1)
            pg.spec.sub_group_policy = [
                {"subGroupSize": 3, "minSubGroups": 2}]
            store.create("PodGroup", pg)
            for i in range(7):      # 2 complete triples + 1 tail
                store.create("Pod", synth.make_pod(
                    f"sg{j}-w-{i}", f"sg{j}", cpu_milli=1000, mem=GI))
        config = default_config()
        config.use_hip = use_hip
        config.device = device
        binder = FakeBinder()
        cache = SchedulerCache(store=store, binder=binder, device=device)
        Scheduler(cache, config).run_once()
        return binder.binds

    cpu = build("cpu", False)
    gpu = build("cuda", True)
    assert len(cpu) == 36               # 6 jobs × 2 triples
    assert cpu == gpu


def test_cycle_equivalence_megacycle(hip, monkeypatch):
    """The opt-in single-launch megacycle path (VAMD_MEGACYCLE=1) makes
    the same decisions as the per-class launch sequence and the CPU
    oracle."""
    monkeypatch.setenv("VAMD_MEGACYCLE", "1")
    from volcano_amd.scheduler import FakeBinder, Scheduler, SchedulerCache, \
        default_config
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth

    GI = 1024 ** 3

    def build(device, use_hip):
        store = ObjectStore()
        for i in range(500):
            store.create("Node", synth.make_node(
                f"n-{i:03d}", cpu_milli=8000, mem=32 * GI))
        store.create("Queue", synth.make_queue("qa", weight=2))
        store.create("Queue", synth.make_queue("qb", weight=1))
        import numpy as np
        rng = np.random.RandomState(11)
        for j in range(60):       # >= 32 classes → megacycle dispatch
            synth.make_gang(store, f"m{j:03d}",
                            replicas=int(rng.randint(1, 5)),
                            queue="qa" if j % 2 else "qb",
                            cpu_milli=float(rng.choice([500, 1000, 1500])),
                            mem=GI, priority=int(rng.randint(0, 3)))
        config = default_config()
        config.use_hip = use_hip
        config.device = device
        binder = FakeBinder()
        cache = SchedulerCache(store=store, binder=binder, device=device)
        sched = Scheduler(cache, config)
        sched.run_once()
        sched.run_once()
        return binder.binds

    cpu = build("cpu", False)
    gpu = build("cuda", True)
    assert len(cpu) > 100
    assert cpu == gpu


def test_cycle_equivalence_chain_path(hip):
    """Chain path (batched score + select-chain with lazy exact re-score)
    vs the torch oracle: a heterogeneous many-small-class inventory
    (hundreds of consecutive single-class jobs — the shape that takes
    the chain path in vamd_run_cycle) must produce IDENTICAL binds.
    Also re-runs with VAMD_NO_CHAIN=1 to confirm the chain changes
    nothing vs the per-class kernel path."""
    import os
    from volcano_amd.scheduler import FakeBinder, Scheduler, SchedulerCache, \
        default_config
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth

    GI = 1024 ** 3

    def build(device, use_hip):
        store = ObjectStore()
        rng = np.random.RandomState(21)
        for i in range(400):
            labels = {"zone": f"z{i % 4}"}
            store.create("Node", synth.make_node(
                f"n-{i:04d}", cpu_milli=float(rng.choice([8000, 16000, 32000])),
                mem=float(rng.choice([16, 32, 64])) * GI, labels=labels))
        for qn, w in (("qa", 3), ("qb", 1), ("qc", 2)):
            store.create("Queue", synth.make_queue(qn, weight=w))
        rng2 = np.random.RandomState(11)
        for j in range(300):
            q = ("qa", "qb", "qc")[j % 3]
            sel = {"zone": f"z{int(rng2.randint(0, 4))}"} \
                if rng2.rand() < 0.25 else None
            synth.make_gang(
                store, f"m{j:03d}", replicas=int(rng2.randint(1, 9)),
                queue=q,
                cpu_milli=float(rng2.choice([250, 500, 1000, 2000])),
                mem=float(rng2.choice([1, 2, 4])) * GI,
                priority=int(rng2.randint(0, 8)),
                node_selector=sel)
        config = default_config()
        config.use_hip = use_hip
        config.device = device
        binder = FakeBinder()
        cache = SchedulerCache(store=store, binder=binder, device=device)
        Scheduler(cache, config).run_once()
        return dict(binder.binds)

    cpu = build("cpu", False)
    os.environ["VAMD_CHAIN_CHUNK"] = "128"   # force the chain at small N
    try:
        gpu_chain = build("cuda", True)
    finally:
        del os.environ["VAMD_CHAIN_CHUNK"]
    assert len(cpu) > 500
    assert cpu == gpu_chain, (
        f"chain path diverges from oracle: {len(cpu)} vs {len(gpu_chain)}; "
        f"sample diff={list(set(cpu.items()) ^ set(gpu_chain.items()))[:6]}")
    os.environ["VAMD_NO_CHAIN"] = "1"
    try:
        gpu_legacy = build("cuda", True)
    finally:
        del os.environ["VAMD_NO_CHAIN"]
    assert gpu_legacy == gpu_chain, "chain vs per-class kernel divergence"


def test_cycle_equivalence_topology_preempt(hip):
    """Topology-aware preemption (domain dry-run trials) produces the
    SAME evictions/pipelines/binds on the CPU oracle and the HIP
    decision plane across a multi-cycle preempt-then-bind sequence."""
    from volcano_amd.api.objects import (HyperNode, HyperNodeMember,
                                         MemberSelector, ObjectMeta)
    from volcano_amd.scheduler import FakeBinder, Scheduler, SchedulerCache, \
        default_config
    from volcano_amd.scheduler.config import PluginOption
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth

    GI = 1024 ** 3

    def mk_hn(name, tier, nodes=None, children=None):
        members = []
        if nodes:
            members.append(HyperNodeMember(
                type="Node", selector=MemberSelector(exact_match=nodes)))
        if children:
            members.append(HyperNodeMember(
                type="HyperNode",
                selector=MemberSelector(exact_match=children)))
        return HyperNode(meta=ObjectMeta(name=name), tier=tier,
                         members=members)

    def build(device, use_hip):
        store = ObjectStore()
        binder = FakeBinder()
        cache = SchedulerCache(store=store, binder=binder, device=device)
        config = default_config()
        config.actions = ["enqueue", "allocate", "preempt", "backfill"]
        config.use_hip = use_hip
        config.device = device
        config.tiers[1].plugins.append(
            PluginOption("network-topology-aware"))
        sched = Scheduler(cache, config)
        for i in range(8):
            store.create("Node", synth.make_node(
                f"n{i}", cpu_milli=4000, mem=16 * GI))
        store.create("HyperNode", mk_hn("rack-a", 1,
                                        nodes=[f"n{i}" for i in range(4)]))
        store.create("HyperNode", mk_hn("rack-b", 1,
                                        nodes=[f"n{i}" for i in range(4, 8)]))
        store.create("HyperNode", mk_hn("spine", 2,
                                        children=["rack-a", "rack-b"]))
        store.create("Queue", synth.make_queue("default"))
        synth.make_gang(store, "filler", replicas=8, min_member=1,
                        cpu_milli=3000, mem=GI, priority=1)
        sched.run_once()
        pg = synth.make_podgroup("net", min_member=3)
        pg.spec.network_topology = {"mode": "hard", "highestTierAllowed": 1}
        store.create("PodGroup", pg)
        for i in range(3):
            store.create("Pod", synth.make_pod(
                f"net-w-{i}", "net", cpu_milli=3000, mem=GI, priority=50))
        sched.run_once()
        evicted = sorted(binder.evictions)
        # let evicted pods terminate, then settle
        for key in evicted:
            ns, name = key.split("/")
            if store.get("Pod", ns, name) is not None:
                store.delete("Pod", ns, name)
        for _ in range(3):
            sched.run_once()
        return evicted, dict(binder.binds)

    ev_cpu, binds_cpu = build("cpu", False)
    ev_gpu, binds_gpu = build("cuda", True)
    assert len(ev_cpu) >= 1
    assert ev_cpu == ev_gpu
    assert binds_cpu == binds_gpu


def test_cycle_equivalence_dra_dims(hip):
    """DRA device-class dims (dra:<class>) ride the dense kernels: the
    CPU oracle and HIP plane make identical placements for claim-backed
    gangs under node device capacity + queue class quotas."""
    from volcano_amd.api.objects import (DeviceClass, ObjectMeta, Queue,
                                         QueueSpec, ResourceClaim)
    from volcano_amd.api.resource import CPU, Resource
    from volcano_amd.scheduler import FakeBinder, Scheduler, SchedulerCache, \
        default_config
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth

    GI = 1024 ** 3

    def build(device, use_hip):
        store = ObjectStore()
        binder = FakeBinder()
        cache = SchedulerCache(store=store, binder=binder, device=device)
        config = default_config()
        config.use_hip = use_hip
        config.device = device
        sched = Scheduler(cache, config)
        store.create("DeviceClass", DeviceClass(
            meta=ObjectMeta(name="mi355x.amd.com"), driver="amdgpu"))
        for i in range(6):
            node = synth.make_node(f"n{i}", cpu_milli=32000, mem=128 * GI)
            node.meta.annotations["dra.volcano.sh/mi355x.amd.com"] = "8"
            store.create("Node", node)
        store.create("Queue", Queue(
            meta=ObjectMeta(name="ml"),
            spec=QueueSpec(weight=1, capability=Resource(
                {CPU: 1e9, "count/mi355x.amd.com": 24.0}))))
        store.create("Queue", synth.make_queue("default"))
        for j in range(8):
            pg = synth.make_podgroup(f"d{j}", queue="ml", min_member=2)
            store.create("PodGroup", pg)
            for i in range(2):
                store.create("ResourceClaim", ResourceClaim(
                    meta=ObjectMeta(name=f"d{j}-{i}", namespace="default"),
                    device_class_name="mi355x.amd.com", count=2))
                pod = synth.make_pod(f"d{j}-w-{i}", f"d{j}", queue="ml",
                                     cpu_milli=1000, mem=GI)
                pod.resource_claims = [f"d{j}-{i}"]
                store.create("Pod", pod)
        sched.run_once()
        return dict(binder.binds)

    cpu = build("cpu", False)
    gpu = build("cuda", True)
    # 24-device quota admits exactly 6 gangs (4 devices each)
    assert len(cpu) == 12
    assert cpu == gpu


@pytest.mark.gpu
@pytest.mark.parametrize("seed", [11, 23, 47])
def test_cycle_equivalence_randomized_sweep(hip, seed):
    """Seeded random inventories (heterogeneous gangs, queues, selectors,
    priorities, partial occupancy) run two cycles on CPU oracle and HIP —
    bind maps must match exactly.  The randomized counterpart of the
    hand-built feature-rich case."""
    from volcano_amd.scheduler import FakeBinder, Scheduler, SchedulerCache, \
        default_config
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth

    GI = 1024 ** 3

    def build(device, use_hip):
        rng = np.random.RandomState(seed)
        store = ObjectStore()
        for i in range(200):
            labels = {"zone": f"z{i % 3}"} if rng.rand() < 0.5 else {}
            store.create("Node", synth.make_node(
                f"n-{i:04d}", cpu_milli=float(rng.choice([4000, 8000,
                                                          16000])),
                mem=float(rng.choice([8, 16, 32])) * GI, labels=labels))
        for qn, w in (("qa", 2), ("qb", 1), ("qc", 3)):
            store.create("Queue", synth.make_queue(qn, weight=w))
        for j in range(60):
            q = ["qa", "qb", "qc"][int(rng.randint(3))]
            size = int(rng.choice([1, 2, 4, 8]))
            sel = {"zone": f"z{int(rng.randint(3))}"} \
                if rng.rand() < 0.25 else None
            synth.make_gang(store, f"r{j:03d}", replicas=size, queue=q,
                            cpu_milli=float(rng.choice([250, 500, 1000,
                                                        2000])),
                            mem=float(rng.choice([1, 2, 4])) * GI,
                            priority=int(rng.randint(0, 8)),
                            node_selector=sel)
        config = default_config()
        config.use_hip = use_hip
        config.device = device
        binder = FakeBinder()
        cache = SchedulerCache(store=store, binder=binder, device=device)
        sched = Scheduler(cache, config)
        sched.run_once()
        # second wave arrives over the partially-occupied cluster
        rng2 = np.random.RandomState(seed + 1)
        for j in range(20):
            synth.make_gang(store, f"w{j:03d}", replicas=int(
                rng2.choice([1, 2, 4])), queue="qb",
                cpu_milli=float(rng2.choice([500, 1000])), mem=GI,
                priority=int(rng2.randint(0, 8)))
        sched.run_once()
        return binder.binds

    cpu = build("cpu", False)
    gpu = build("cuda", True)
    assert len(cpu) > 100
    assert cpu == gpu, (
        f"decision divergence at seed {seed}: {len(cpu)} vs {len(gpu)}; "
        f"diff={sorted(set(cpu.items()) ^ set(gpu.items()))[:6]}")


@pytest.mark.gpu
def test_cycle_equivalence_hdrf_caps(hip):
    """Hierarchical-DRF equilibrium caps feed the in-kernel queue clamp:
    CPU oracle and HIP must bind identically under enableHierarchy."""
    from volcano_amd.scheduler import FakeBinder, Scheduler, SchedulerCache, \
        default_config
    from volcano_amd.scheduler.config import PluginOption
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth

    GI = 1024 ** 3
    HIER = "volcano.sh/hierarchy"
    HIERW = "volcano.sh/hierarchy-weights"

    def build(device, use_hip):
        store = ObjectStore()
        for i in range(50):
            store.create("Node", synth.make_node(
                f"n-{i:03d}", cpu_milli=8000, mem=16 * GI))
        for qn, h, w in (("sci", "root/sci", "100/50"),
                         ("dev", "root/eng/dev", "100/50/30"),
                         ("prod", "root/eng/prod", "100/50/70")):
            q = synth.make_queue(qn)
            q.meta.annotations[HIER] = h
            q.meta.annotations[HIERW] = w
            store.create("Queue", q)
        rng = np.random.RandomState(7)
        for j in range(45):
            qn = ("sci", "dev", "prod")[j % 3]
            synth.make_gang(store, f"h{j:03d}",
                            replicas=int(rng.choice([2, 4, 8])), queue=qn,
                            cpu_milli=float(rng.choice([500, 1000, 2000])),
                            mem=float(rng.choice([1, 2])) * GI,
                            priority=int(rng.randint(0, 4)))
        config = default_config()
        config.use_hip = use_hip
        config.device = device
        for tier in config.tiers:
            for p in tier.plugins:
                if p.name == "drf":
                    p.arguments = {"enableHierarchy": True}
        binder = FakeBinder()
        cache = SchedulerCache(store=store, binder=binder, device=device)
        sched = Scheduler(cache, config)
        sched.run_once()
        sched.run_once()
        return binder.binds

    cpu = build("cpu", False)
    gpu = build("cuda", True)
    assert len(cpu) > 50
    assert cpu == gpu, (
        f"HDRF decision divergence: {len(cpu)} vs {len(gpu)}; "
        f"diff={sorted(set(cpu.items()) ^ set(gpu.items()))[:6]}")
