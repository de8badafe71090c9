"""Pure-PyTorch reference implementations of the scheduler kernels.

These define the *semantics* the HIP kernels (``csrc/scheduler_kernels.hip``)
must reproduce exactly on the decision level (same placements, same ready
flags).  They run on CPU for the non-GPU test tier and are the golden
oracle for the GPU numerics tests.

Design rule — **sync-free hot loop**: every op is buffer-style (writes into
caller-provided device tensors, returns nothing host-visible).  The per-job
gang decision (commit vs revert) is made *on device* so a whole scheduling
cycle is one enqueue of kernels on one HIP stream with a single D2H readback
at the end.  The reference (Go) makes the same decisions with per-task
host code — the order-equivalent semantics are documented per op.

Kernel inventory (SURVEY.md §2.9 K1–K6 mapping):

* ``score_cap``      — K1+K2 fused: feasibility + capacity + score per node.
* ``select_commit``  — K2 select + K4 gang placement with undo log.
* ``finalize_job``   — K4 gang readiness decision (device flag).
* ``cond_revert``    — K4 rollback of one class, gated on the job flag.
* ``drf_share``      — K3 dominant share per job.
* ``waterfill``      — K3 proportion/capacity deserved water-filling.

Decision determinism: score ties break toward the lower node index
(everywhere — CPU reference, HIP kernels, torch fallbacks).
"""

from __future__ import annotations

from typing import Optional

import torch

NEG_INF = float("-inf")
EPS = 1e-4          # resource-fit slack, mirrors api/resource.MIN_RESOURCE scale
BIG_CAP = 2 ** 30   # "unbounded" capacity sentinel


def score_cap(
    alloc: torch.Tensor,        # [N, R] f32 allocatable
    used: torch.Tensor,         # [N, R] f32 currently used (incl. staged)
    extra: torch.Tensor,        # [N, R] f32 future credit (releasing − pipelined); zeros for idle-only
    ready: torch.Tensor,        # [N] bool node schedulable
    taints: torch.Tensor,       # [N] i64 taint bitmask
    planes: torch.Tensor,       # [N, W] i64 label planes
    req: torch.Tensor,          # [R] f32 per-task request
    tolerated: int,             # i64 taint bits this class tolerates
    require: torch.Tensor,      # [W] i64 plane bits that must all be set
    forbid: torch.Tensor,       # [W] i64 plane bits that must all be clear
    w_least: float,             # leastrequested weight (nodeorder)
    w_most: float,              # mostrequested / binpack weight
    w_bal: float,               # balancedallocation weight
    dim_w: torch.Tensor,        # [R] f32 per-resource weights (binpack.weight.<res>)
    bias: Optional[torch.Tensor],  # [N] f32 additive score bias (or None)
    score_out: torch.Tensor,    # [N] f32 out: -inf for infeasible
    cap_out: torch.Tensor,      # [N] i32 out: instances that fit
) -> None:
    avail = alloc - used + extra                              # [N, R]

    fits = (avail + EPS >= req).all(dim=1)
    taint_ok = (taints & ~torch.tensor(tolerated, dtype=torch.int64, device=taints.device)) == 0
    aff_ok = ((planes & require) == require).all(dim=1)
    anti_ok = ((planes & forbid) == 0).all(dim=1)
    feasible = ready & fits & taint_ok & aff_ok & anti_ok

    pos = req > EPS
    per_dim = torch.where(pos, (avail + EPS) / torch.clamp(req, min=EPS),
                          torch.full_like(avail, float(BIG_CAP)))
    cap = per_dim.amin(dim=1).clamp(min=0, max=float(BIG_CAP)).floor().to(torch.int32)
    cap_out.copy_(torch.where(feasible, cap, torch.zeros_like(cap)))

    denom = torch.clamp(alloc, min=EPS)
    frac = torch.clamp((used + req) / denom, max=1.0)         # [N, R]
    wsum = torch.clamp(dim_w.sum(), min=EPS)
    least = ((1.0 - frac) * dim_w).sum(dim=1) / wsum
    most = (frac * dim_w).sum(dim=1) / wsum
    mean = frac.mean(dim=1, keepdim=True)
    bal = 1.0 - torch.sqrt(((frac - mean) ** 2).mean(dim=1))
    score = w_least * least + w_most * most + w_bal * bal
    if bias is not None:
        score = score + bias
    score_out.copy_(torch.where(feasible & (cap_out > 0), score,
                                torch.full_like(score, NEG_INF)))


def select_commit(
    score: torch.Tensor,        # [N] f32 from score_cap
    cap: torch.Tensor,          # [N] i32
    req: torch.Tensor,          # [R] f32
    ntasks: int,                # class size (instances to place)
    used: torch.Tensor,         # [N, R] f32 — updated in place (staged)
    queue_alloc: torch.Tensor,  # [R] f32 — this queue's allocated row, in place
    queue_limit: torch.Tensor,  # [R] f32 — queue bound (+inf = none)
    log_nodes: torch.Tensor,    # [K] i32 — this class's undo-log slot (K ≥ min(ntasks, N))
    log_counts: torch.Tensor,   # [K] i32
    log_len: torch.Tensor,      # [] i32 — entries written
    placed: torch.Tensor,       # [] i32 — this class's placed count
    job_placed: torch.Tensor,   # [] i32 — accumulator across the job's classes
) -> None:
    """Greedy best-score fill for one task class.

    Sequential-equivalent semantics: repeatedly take the feasible node with
    the highest score (ties → lowest index), place as many instances as its
    capacity and the queue quota allow, move on.  The score is *not*
    recomputed between placements within one class — all instances of a
    class are identical, so the node order is static and the fill equals
    "sort by score desc, take cap each".  (The reference scores per task and
    samples 5 % of nodes — predicate_helper.go:45, scheduler_helper.go:56;
    we score ALL nodes, which strictly improves placement quality.)
    """
    N = score.shape[0]
    log_nodes.zero_()
    log_counts.zero_()

    # queue quota → instance budget under the queue limit
    headroom = queue_limit - queue_alloc                      # [R]
    pos = req > EPS
    per_dim = torch.where(pos, (headroom + EPS) / torch.clamp(req, min=EPS),
                          torch.full_like(headroom, float(BIG_CAP)))
    quota = per_dim.amin().clamp(min=0, max=float(BIG_CAP)).floor()
    budget = torch.minimum(torch.tensor(float(ntasks), device=score.device), quota)

    if N == 0:
        log_len.zero_(); placed.zero_()
        return
    order = torch.argsort(score, descending=True, stable=True)
    sorted_cap = torch.where(score[order] > NEG_INF, cap[order],
                             torch.zeros_like(cap)).to(torch.int64)
    csum = torch.cumsum(sorted_cap, dim=0)
    take = torch.clamp(sorted_cap - torch.clamp(csum - budget.to(torch.int64), min=0), min=0)
    total = torch.minimum(csum[-1], budget.to(torch.int64))

    k = log_nodes.shape[0]
    # compact the first k nonzero entries into the log slot
    # (torch.nonzero syncs on GPU — acceptable: the GPU path runs the HIP
    #  kernel, which compacts in-kernel; this reference is the CPU oracle)
    idx = torch.nonzero(take > 0, as_tuple=False).flatten()[:k]
    cnt = take[idx].to(torch.int32)
    m = int(idx.shape[0])
    log_nodes[:m] = order[idx].to(torch.int32)
    log_counts[:m] = cnt
    log_len.fill_(m)

    used.index_add_(0, order[idx].to(torch.long),
                    cnt.to(torch.float32).unsqueeze(1) * req.unsqueeze(0))
    queue_alloc += total.to(torch.float32) * req
    placed.copy_(total.to(torch.int32))
    job_placed.add_(total.to(torch.int32))


def finalize_job(
    job_placed: torch.Tensor,   # [] i32 placed across the job's classes
    occupied: int,              # tasks already holding resources (host-known at cycle start)
    min_available: int,         # gang minimum
    class_placed: torch.Tensor,  # [nc] i32 per-class placed counts
    class_min: torch.Tensor,     # [nc] i32 per-class minimums (role mins)
    flag: torch.Tensor,         # [] u8 out: 1 = commit, 0 = revert
) -> None:
    """Gang readiness (session_plugins.go:483 JobReady AND-aggregation)."""
    ok = (job_placed + occupied >= min_available)
    ok = ok & (class_placed >= class_min).all()
    flag.copy_(ok.to(torch.uint8))


def cond_revert(
    flag: torch.Tensor,         # [] u8 — 1 = keep, 0 = revert
    log_nodes: torch.Tensor,    # [K] i32
    log_counts: torch.Tensor,   # [K] i32 — zeroed on revert
    log_len: torch.Tensor,      # [] i32
    req: torch.Tensor,          # [R] f32
    used: torch.Tensor,         # [N, R] f32 in place
    queue_alloc: torch.Tensor,  # [R] f32 in place
    placed: torch.Tensor,       # [] i32 — zeroed on revert
    job_placed: torch.Tensor,   # [] i32 — decremented on revert
) -> None:
    """Undo one class's staged placement iff the gang failed
    (statement.go:375 Discard, reverse-order undo)."""
    do = (flag == 0).to(torch.float32)                        # 0.0 keep / 1.0 revert
    cnt = log_counts.to(torch.float32) * do
    arange = torch.arange(log_counts.shape[0], device=log_counts.device)
    valid = arange < log_len.to(arange.dtype)
    cnt = cnt * valid.to(torch.float32)
    used.index_add_(0, log_nodes.to(torch.long),
                    -cnt.unsqueeze(1) * req.unsqueeze(0))
    queue_alloc -= cnt.sum() * req
    job_placed.add_((-placed * (flag == 0).to(torch.int32)))
    placed.mul_((flag != 0).to(torch.int32))
    log_counts.mul_((flag != 0).to(torch.int32))


def drf_share(job_alloc: torch.Tensor, total: torch.Tensor) -> torch.Tensor:
    """Dominant share per job: max_r alloc/total (drf.go calculateShare).

    job_alloc [J, R], total [R] → [J] f64."""
    t = torch.clamp(total.to(torch.float64), min=1.0)
    return (job_alloc.to(torch.float64) / t).amax(dim=1)


def waterfill(
    weight: torch.Tensor,       # [Q] f32 queue weights
    request: torch.Tensor,      # [Q, R] f32 total demand per queue
    guarantee: torch.Tensor,    # [Q, R] f32 floor
    capability: torch.Tensor,   # [Q, R] f32 ceiling (+inf = none)
    total: torch.Tensor,        # [R] f32 cluster total
    iters: int = 16,
) -> torch.Tensor:
    """Proportion-plugin deserved computation (proportion.go:90-260).

    Iterative weighted water-filling: each round splits the remaining pool
    by weight among unsatisfied queues, capping at min(max(request,
    guarantee), capability); guarantees are granted first.  float64
    accumulators for reproducibility (SURVEY §7 hard-parts note)."""
    w = weight.to(torch.float64)
    demand = torch.minimum(request.to(torch.float64), capability.to(torch.float64))
    demand = torch.maximum(demand, guarantee.to(torch.float64))
    deserved = torch.minimum(guarantee.to(torch.float64),
                             capability.to(torch.float64)).clone()
    remaining = torch.clamp(total.to(torch.float64) - deserved.sum(dim=0), min=0.0)

    for _ in range(iters):
        unsat = deserved + 1e-9 < demand                      # [Q, R]
        wsum = (w.unsqueeze(1) * unsat).sum(dim=0)            # [R]
        give = torch.where(wsum > 0, remaining / torch.clamp(wsum, min=1e-12),
                           torch.zeros_like(remaining))
        inc = torch.minimum(w.unsqueeze(1) * give.unsqueeze(0) * unsat,
                            demand - deserved)
        inc = torch.clamp(inc, min=0.0)
        deserved = deserved + inc
        remaining = torch.clamp(remaining - inc.sum(dim=0), min=0.0)
    return deserved.to(torch.float32)
