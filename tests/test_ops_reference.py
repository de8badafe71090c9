"""Torch-oracle kernel semantics (volcano_amd/ops/reference.py).

These pin the *decision* semantics the HIP kernels must reproduce; the
GPU numerics tests (test_ops_gpu.py) compare HIP against these exact
functions.
"""

import torch

from volcano_amd.ops import reference as ref


def _mk_state(N=8, R=3):
    alloc = torch.full((N, R), 10.0)
    used = torch.zeros(N, R)
    extra = torch.zeros(N, R)
    ready = torch.ones(N, dtype=torch.bool)
    taints = torch.zeros(N, dtype=torch.int64)
    planes = torch.zeros(N, 1, dtype=torch.int64)
    return alloc, used, extra, ready, taints, planes


def _score(alloc, used, extra, ready, taints, planes, req, tol=-1,
           require=None, forbid=None, w=(1.0, 0.0, 0.0), dim_w=None, bias=None):
    N, R = alloc.shape
    W = planes.shape[1]
    require = require if require is not None else torch.zeros(W, dtype=torch.int64)
    forbid = forbid if forbid is not None else torch.zeros(W, dtype=torch.int64)
    dim_w = dim_w if dim_w is not None else torch.ones(R)
    score = torch.empty(N)
    cap = torch.empty(N, dtype=torch.int32)
    ref.score_cap(alloc, used, extra, ready, taints, planes, req, tol,
                  require, forbid, w[0], w[1], w[2], dim_w, bias, score, cap)
    return score, cap


def test_score_cap_fit_and_capacity():
    alloc, used, extra, ready, taints, planes = _mk_state()
    used[0, 0] = 9.5           # node 0 nearly full on dim 0
    ready[1] = False           # node 1 unschedulable
    req = torch.tensor([2.0, 1.0, 0.0])
    score, cap = _score(alloc, used, extra, ready, taints, planes, req)
    assert score[0] == float("-inf") and cap[0] == 0
    assert score[1] == float("-inf") and cap[1] == 0
    assert cap[2] == 5          # 10/2
    assert score[2] > float("-inf")


def test_score_cap_taints_and_planes():
    alloc, used, extra, ready, taints, planes = _mk_state()
    taints[3] = 0b10
    planes[4, 0] = 0b100       # node 4 has label bit 2
    req = torch.tensor([1.0, 0.0, 0.0])
    # intolerant task: node 3 infeasible
    score, _ = _score(alloc, used, extra, ready, taints, planes, req, tol=0)
    assert score[3] == float("-inf")
    # tolerant task
    score, _ = _score(alloc, used, extra, ready, taints, planes, req, tol=0b10)
    assert score[3] > float("-inf")
    # require bit 2 → only node 4 feasible
    score, _ = _score(alloc, used, extra, ready, taints, planes, req,
                      require=torch.tensor([0b100]))
    ok = score > float("-inf")
    assert ok[4] and ok.sum() == 1
    # forbid bit 2 → node 4 infeasible
    score, _ = _score(alloc, used, extra, ready, taints, planes, req,
                      forbid=torch.tensor([0b100]))
    assert score[4] == float("-inf") and (score > float("-inf")).sum() == 7


def test_score_least_requested_prefers_empty():
    alloc, used, extra, ready, taints, planes = _mk_state()
    used[2] = 5.0              # node 2 half full
    req = torch.tensor([1.0, 1.0, 1.0])
    score, _ = _score(alloc, used, extra, ready, taints, planes, req,
                      w=(1.0, 0.0, 0.0))
    assert score[0] > score[2]
    # binpack (most) prefers the fuller node
    score, _ = _score(alloc, used, extra, ready, taints, planes, req,
                      w=(0.0, 1.0, 0.0))
    assert score[2] > score[0]


def _mk_commit(N=8, R=2, ntasks=5, qlimit=None):
    alloc = torch.full((N, R), 4.0)
    used = torch.zeros(N, R)
    score = torch.linspace(1.0, 0.1, N)           # node 0 best
    cap = torch.full((N,), 2, dtype=torch.int32)  # 2 instances per node
    req = torch.tensor([1.0, 1.0])
    queue_alloc = torch.zeros(R)
    queue_limit = qlimit if qlimit is not None else torch.full((R,), 1e18)
    K = min(ntasks, N)
    log_nodes = torch.zeros(K, dtype=torch.int32)
    log_counts = torch.zeros(K, dtype=torch.int32)
    log_len = torch.zeros((), dtype=torch.int32)
    placed = torch.zeros((), dtype=torch.int32)
    job_placed = torch.zeros((), dtype=torch.int32)
    ref.select_commit(score, cap, req, ntasks, used, queue_alloc, queue_limit,
                      log_nodes, log_counts, log_len, placed, job_placed)
    return dict(used=used, queue_alloc=queue_alloc, log_nodes=log_nodes,
                log_counts=log_counts, log_len=log_len, placed=placed,
                job_placed=job_placed, req=req, score=score)


def test_select_commit_greedy_fill():
    st = _mk_commit(ntasks=5)
    # best 3 nodes by score: 0,1,2 with caps 2,2,1
    assert int(st["placed"]) == 5
    assert st["log_nodes"][: int(st["log_len"])].tolist() == [0, 1, 2]
    assert st["log_counts"][: int(st["log_len"])].tolist() == [2, 2, 1]
    assert st["used"][0, 0] == 2.0 and st["used"][2, 0] == 1.0
    assert st["queue_alloc"].tolist() == [5.0, 5.0]


def test_select_commit_queue_quota():
    st = _mk_commit(ntasks=5, qlimit=torch.tensor([3.0, 1e18]))
    assert int(st["placed"]) == 3       # quota caps at 3 instances
    assert st["queue_alloc"][0] == 3.0


def test_gang_revert_roundtrip():
    st = _mk_commit(ntasks=5)
    flag = torch.zeros((), dtype=torch.uint8)   # revert
    ref.cond_revert(flag, st["log_nodes"], st["log_counts"], st["log_len"],
                    st["req"], st["used"], st["queue_alloc"], st["placed"],
                    st["job_placed"])
    assert st["used"].abs().sum() == 0
    assert st["queue_alloc"].abs().sum() == 0
    assert int(st["placed"]) == 0 and int(st["job_placed"]) == 0
    assert st["log_counts"].abs().sum() == 0


def test_finalize_job_and_keep():
    st = _mk_commit(ntasks=5)
    flag = torch.zeros((), dtype=torch.uint8)
    class_placed = torch.tensor([5], dtype=torch.int32)
    class_min = torch.tensor([3], dtype=torch.int32)
    ref.finalize_job(st["job_placed"], 0, 5, class_placed, class_min, flag)
    assert int(flag) == 1
    ref.finalize_job(st["job_placed"], 0, 6, class_placed, class_min, flag)
    assert int(flag) == 0
    # role minimum not met
    ref.finalize_job(st["job_placed"], 0, 5, class_placed,
                     torch.tensor([6], dtype=torch.int32), flag)
    assert int(flag) == 0


def test_drf_share():
    alloc = torch.tensor([[10.0, 0.0], [5.0, 40.0]])
    total = torch.tensor([100.0, 100.0])
    s = ref.drf_share(alloc, total)
    assert abs(float(s[0]) - 0.10) < 1e-9
    assert abs(float(s[1]) - 0.40) < 1e-9


def test_waterfill_weighted_split():
    # two queues weight 2:1, plenty of demand, total 90 → 60/30
    weight = torch.tensor([2.0, 1.0])
    request = torch.full((2, 1), 1000.0)
    guarantee = torch.zeros(2, 1)
    capability = torch.full((2, 1), 1e18)
    total = torch.tensor([90.0])
    d = ref.waterfill(weight, request, guarantee, capability, total)
    assert abs(float(d[0, 0]) - 60.0) < 1e-3
    assert abs(float(d[1, 0]) - 30.0) < 1e-3


def test_waterfill_capped_redistributes():
    # q0 capped at 10 → q1 gets the rest
    weight = torch.tensor([1.0, 1.0])
    request = torch.full((2, 1), 1000.0)
    guarantee = torch.zeros(2, 1)
    capability = torch.tensor([[10.0], [1e18]])
    total = torch.tensor([100.0])
    d = ref.waterfill(weight, request, guarantee, capability, total)
    assert abs(float(d[0, 0]) - 10.0) < 1e-3
    assert abs(float(d[1, 0]) - 90.0) < 1e-3


def test_waterfill_guarantee_floor():
    weight = torch.tensor([1.0, 1.0])
    request = torch.tensor([[0.0], [1000.0]])   # q0 requests nothing
    guarantee = torch.tensor([[20.0], [0.0]])   # but is guaranteed 20
    capability = torch.full((2, 1), 1e18)
    total = torch.tensor([100.0])
    d = ref.waterfill(weight, request, guarantee, capability, total)
    assert float(d[0, 0]) >= 20.0 - 1e-6
    assert abs(float(d[1, 0]) - 80.0) < 1e-3
