"""Cycle plan: the host-built launch schedule for one allocate pass.

The reference's allocate action walks queues→jobs→tasks with Go callbacks
per (task, node) (`actions/allocate/allocate.go:719-866`).  Here the host
builds a *plan* — an ordered array of task-class descriptors and job gang
descriptors — and the whole cycle executes as one enqueue:

* GPU path: a single ``vamd_run_cycle`` library call replays the plan as
  back-to-back HIP kernels on one stream, zero host syncs mid-cycle
  (``ops/csrc/cycle_runner.hip``); one D2H readback returns every
  placement.
* CPU path: the same plan interpreted over the torch oracle ops
  (``ops/reference.py``) — bit-identical decisions, used by the non-GPU
  test tier and as the golden reference for the HIP kernels.
"""

from __future__ import annotations

import ctypes
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from ..api.info import JobInfo, TaskClass
from ..ops import reference as ref
from .tensors import NodeTensors

BIG_LIMIT = 1.0e18     # "no queue limit" sentinel (finite: kernel does int cast)

# binary-compatible numpy mirrors of VamdClassDesc / VamdJobDesc
# (ops/csrc/vamd_api.h) — the descriptor marshalling fills these columns
# instead of per-field ctypes stores (which cost ~10 µs per class)
CLASS_DT = np.dtype([
    ("job_idx", "<i4"), ("queue_idx", "<i4"), ("ntasks", "<i4"),
    ("min_needed", "<i4"), ("log_off", "<i4"), ("log_cap", "<i4"),
    ("flags", "<i4"), ("bias_row", "<i4"),
    ("w_least", "<f4"), ("w_most", "<f4"), ("w_bal", "<f4"),
    ("_padf", "<f4")])
JOB_DT = np.dtype([
    ("class_begin", "<i4"), ("class_end", "<i4"),
    ("occupied", "<i4"), ("min_available", "<i4")])


@dataclass(slots=True)
class BundleEntry:
    """One gang job inside a fused bundle class (see ClassPlan.bundle)."""
    job_key: str
    tasks: list                # the job's pending TaskInfos (class order)
    ntasks: int
    min_needed: int            # gang minimum for THIS job
    job: object = None         # JobInfo ref (skips the apply-time lookup)


@dataclass(slots=True)
class ClassPlan:
    tclass: TaskClass
    job_key: str
    queue_idx: int
    req: np.ndarray            # [R] f32
    tolerated: int
    require: np.ndarray        # [W] i64
    forbid: np.ndarray         # [W] i64
    min_needed: int            # per-role minimum still missing
    w_least: float = 1.0
    w_most: float = 0.0
    w_bal: float = 0.0
    use_future: bool = False   # score against future-idle (pipelining)
    # optional per-class additive score plane [N] f32 (task-topology
    # bucket packing); overrides the plan-wide bias for this class —
    # builders that want both must pre-add plan.bias into the row
    bias: Optional[np.ndarray] = None
    log_off: int = 0
    log_cap: int = 0
    # instance count override (bundles track the sum here instead of
    # materializing a merged 100k-task list; tclass keeps ONE
    # representative task for constraint/handler access)
    ntasks_override: Optional[int] = None
    # Gang bundling: consecutive jobs with identical single-class shape are
    # fused into ONE kernel pass (ntasks = sum); the per-job gang
    # boundaries live here and are enforced at apply time (slot walk with
    # recycling).  The kernel-side fuse_min is the EASIEST entry's
    # minimum: if even that gang can't place, the bundle reverts in-kernel.
    bundle: Optional[List[BundleEntry]] = None

    @property
    def ntasks(self) -> int:
        return self.ntasks_override if self.ntasks_override is not None \
            else self.tclass.count


@dataclass(slots=True)
class JobPlan:
    job_key: str
    class_begin: int
    class_end: int
    occupied: int
    min_available: int


@dataclass
class ClassResult:
    placements: List[Tuple[int, int]]   # (node_id, count)
    placed: int


class CycleResult:
    """Array-backed cycle outcome (one readback): per-class undo logs,
    placed counts, per-job commit flags.  The apply path consumes the
    numpy arrays directly; the dict/object views materialize lazily for
    cold consumers (subgroup gating, tests)."""

    def __init__(self, plan, log_nodes, log_counts, log_len, class_placed,
                 job_placed, job_flag):
        self.plan = plan
        self.log_nodes = log_nodes
        self.log_counts = log_counts
        self.log_len = log_len
        self.class_placed = class_placed
        self.job_placed_arr = job_placed
        self.job_flag = job_flag
        self._committed = None
        self._class_results = None

    @property
    def job_committed(self) -> Dict[str, bool]:
        if self._committed is None:
            self._committed = {jp.job_key: bool(self.job_flag[j])
                               for j, jp in enumerate(self.plan.jobs)}
        return self._committed

    @property
    def job_placed(self) -> Dict[str, int]:
        return {jp.job_key: int(self.job_placed_arr[j])
                for j, jp in enumerate(self.plan.jobs)}

    def class_entries(self, c: int):
        """(nodes, counts) numpy views of class c's undo log."""
        cp = self.plan.classes[c]
        n = int(self.log_len[c])
        sl = slice(cp.log_off, cp.log_off + n)
        return self.log_nodes[sl], self.log_counts[sl]

    @property
    def class_results(self) -> List[ClassResult]:
        if self._class_results is None:
            out = []
            for c in range(len(self.plan.classes)):
                ln, lc = self.class_entries(c)
                entries = [(int(n), int(k)) for n, k in zip(ln, lc)
                           if k > 0]
                out.append(ClassResult(entries, int(self.class_placed[c])))
            self._class_results = out
        return self._class_results


class CyclePlan:
    """Ordered classes + gang jobs + packed per-class constraint tensors."""

    def __init__(self, nt: NodeTensors, queue_limit: torch.Tensor,
                 queue_alloc: torch.Tensor):
        self.nt = nt
        self.classes: List[ClassPlan] = []
        self.jobs: List[JobPlan] = []
        self.queue_limit = queue_limit          # [Q, R] f32 (host)
        self.queue_alloc = queue_alloc          # [Q, R] f32 (host)
        self.bias: Optional[torch.Tensor] = None
        self.dim_w = torch.ones(nt.r, dtype=torch.float32)
        self.log_total = 0

    def add_job(self, job: JobInfo, classes: List[ClassPlan]) -> None:
        if not classes:
            return
        begin = len(self.classes)
        self.classes.extend(classes)
        self.jobs.append(JobPlan(
            job_key=job.key, class_begin=begin, class_end=len(self.classes),
            occupied=job.occupied_count, min_available=job.min_available))

    def finalize(self) -> None:
        """Assign undo-log slots (one contiguous region per class),
        normalize constraint widths (the label-bit registry may have
        grown while classes were built), and stage the per-class job
        index + request matrix both runners and the apply path share."""
        # per-class bias rows REPLACE the plan bias in the kernel — fold
        # the plan-wide plane (score_bias / soft-shard stagger) into them
        # so biased classes don't lose it.  Shared rows fold once.
        if self.bias is not None:
            pb = self.bias.cpu().numpy().astype(np.float32)
            originals = [cp.bias for cp in self.classes
                         if cp.bias is not None]     # pin ids during fold
            folded: Dict[int, np.ndarray] = {}
            for cp in self.classes:
                if cp.bias is not None:
                    nb = folded.get(id(cp.bias))
                    if nb is None:
                        nb = folded[id(cp.bias)] = \
                            (cp.bias + pb).astype(np.float32)
                    cp.bias = nb
            del originals
        self.nt.ensure_plane_width()
        W = self.nt.planes_t.shape[0] if self.nt.planes_t is not None \
            else max(self.nt.labels.words, 1)
        off = 0
        n = self.nt.n
        C = len(self.classes)
        self.req_np = np.empty((C, self.nt.r), dtype=np.float32)
        for c, cp in enumerate(self.classes):
            cp.log_off = off
            cp.log_cap = max(1, min(cp.ntasks, n))
            off += cp.log_cap
            if len(cp.require) < W:
                cp.require = np.pad(cp.require, (0, W - len(cp.require)))
            if len(cp.forbid) < W:
                cp.forbid = np.pad(cp.forbid, (0, W - len(cp.forbid)))
            self.req_np[c] = cp.req
        self.log_total = off
        lens = np.fromiter((jp.class_end - jp.class_begin
                            for jp in self.jobs), dtype=np.int64,
                           count=len(self.jobs))
        self.class_job = np.repeat(
            np.arange(len(self.jobs), dtype=np.int32), lens)

    @property
    def n_classes(self) -> int:
        return len(self.classes)


def run_plan_torch(plan: CyclePlan) -> CycleResult:
    """Interpret the plan over the torch oracle ops — mirrors
    ``cycle_runner.hip`` statement for statement (CPU test tier)."""
    nt = plan.nt
    N, R = nt.n, nt.r
    dev = nt.alloc_t.device
    C = plan.n_classes
    if C == 0:
        return CycleResult([], {}, {})

    alloc = nt.alloc_t.t()          # [N, R] views over the [R, N] planes
    used = nt.used_t.t()
    extra_full = nt.extra_t.t()
    zeros_extra = torch.zeros_like(extra_full)
    planes = nt.planes_t.t()        # [N, W]

    dim_w = plan.dim_w.to(dev)
    score = torch.empty(N, dtype=torch.float32, device=dev)
    cap = torch.empty(N, dtype=torch.int32, device=dev)
    log_nodes = torch.zeros(plan.log_total, dtype=torch.int32, device=dev)
    log_counts = torch.zeros(plan.log_total, dtype=torch.int32, device=dev)
    log_len = torch.zeros(C, dtype=torch.int32, device=dev)
    class_placed = torch.zeros(C, dtype=torch.int32, device=dev)
    job_placed = torch.zeros(len(plan.jobs), dtype=torch.int32, device=dev)
    job_flag = torch.ones(len(plan.jobs), dtype=torch.uint8, device=dev)
    class_min = torch.tensor([cp.min_needed for cp in plan.classes],
                             dtype=torch.int32, device=dev)

    for j, jp in enumerate(plan.jobs):
        nc = jp.class_end - jp.class_begin
        single = nc == 1
        for c in range(jp.class_begin, jp.class_end):
            cp = plan.classes[c]
            req = torch.from_numpy(cp.req).to(dev)
            require = torch.from_numpy(cp.require).to(dev)
            forbid = torch.from_numpy(cp.forbid).to(dev)
            extra = extra_full if cp.use_future else zeros_extra
            b = plan.bias if cp.bias is None \
                else torch.from_numpy(cp.bias).to(dev)
            ref.score_cap(alloc, used, extra, nt.ready.bool(), nt.taint_mask,
                          planes, req, cp.tolerated, require, forbid,
                          cp.w_least, cp.w_most, cp.w_bal, dim_w, b,
                          score, cap)
            sl = slice(cp.log_off, cp.log_off + cp.log_cap)
            ref.select_commit(score, cap, req, cp.ntasks, used,
                              plan.queue_alloc[cp.queue_idx],
                              plan.queue_limit[cp.queue_idx],
                              log_nodes[sl], log_counts[sl], log_len[c],
                              class_placed[c], job_placed[j])
            if single:
                need = max(jp.min_available - jp.occupied, cp.min_needed, 0)
                flag = (class_placed[c] >= need).to(torch.uint8)
                job_flag[j] = flag
                if int(flag) == 0:
                    ref.cond_revert(flag, log_nodes[sl], log_counts[sl],
                                    log_len[c], req, used,
                                    plan.queue_alloc[cp.queue_idx],
                                    class_placed[c], job_placed[j])
        if not single:
            ref.finalize_job(job_placed[j], jp.occupied, jp.min_available,
                             class_placed[jp.class_begin:jp.class_end],
                             class_min[jp.class_begin:jp.class_end],
                             job_flag[j])
            for c in range(jp.class_begin, jp.class_end):
                cp = plan.classes[c]
                req = torch.from_numpy(cp.req).to(dev)
                sl = slice(cp.log_off, cp.log_off + cp.log_cap)
                ref.cond_revert(job_flag[j], log_nodes[sl], log_counts[sl],
                                log_len[c], req, used,
                                plan.queue_alloc[cp.queue_idx],
                                class_placed[c], job_placed[j])

    return CycleResult(plan, log_nodes.cpu().numpy(),
                      log_counts.cpu().numpy(), log_len.cpu().numpy(),
                      class_placed.cpu().numpy(), job_placed.cpu().numpy(),
                      job_flag.cpu().numpy())


def run_plan_hip(plan: CyclePlan) -> CycleResult:
    """One library call → whole cycle on the GPU (no host syncs mid-cycle)."""
    from ..ops import hip

    nt = plan.nt
    N, R = nt.n, nt.r
    W = nt.planes_t.shape[0]
    dev = nt.alloc_t.device
    C = plan.n_classes
    if C == 0:
        return CycleResult([], {}, {})
    if R > 64:
        # revert/commit kernels map one wavefront (64 lanes) per log
        # entry — more resource dims than lanes would silently skip dims
        raise RuntimeError(
            f"HIP decision plane supports up to 64 resource dims, got {R} "
            "(synthetic paa:/hp: dims count); split the inventory or use "
            "the torch oracle path")

    # ---- descriptor marshalling: ONE pass over the classes filling
    # numpy staging buffers (ctypes per-field struct stores measured
    # ~100 ms/cycle on 10k-class mixed plans — the numpy columns are
    # binary-compatible with VamdClassDesc/VamdJobDesc).
    J = len(plan.jobs)
    cds = np.empty(C, dtype=CLASS_DT)
    jds = np.empty(J, dtype=JOB_DT)
    for j, jp in enumerate(plan.jobs):
        jds[j] = (jp.class_begin, jp.class_end, jp.occupied,
                  jp.min_available)
    job_idx = plan.class_job

    req_np = plan.req_np
    require_np = np.empty((C, W), dtype=np.int64)
    forbid_np = np.empty((C, W), dtype=np.int64)
    tol_np = np.empty(C, dtype=np.int64)
    bias_rows_list: List[np.ndarray] = []
    bias_row_of: Dict[int, int] = {}      # id(array) -> row index
    for c, cp in enumerate(plan.classes):
        if cp.bias is None:
            brow = -1
        else:
            brow = bias_row_of.get(id(cp.bias))
            if brow is None:
                brow = bias_row_of[id(cp.bias)] = len(bias_rows_list)
                bias_rows_list.append(cp.bias)
        nt_override = cp.ntasks_override
        cds[c] = (job_idx[c], cp.queue_idx,
                  nt_override if nt_override is not None
                  else len(cp.tclass.tasks),
                  cp.min_needed, cp.log_off, cp.log_cap,
                  1 if cp.use_future else 0, brow,
                  cp.w_least, cp.w_most, cp.w_bal, 0.0)
        require_np[c] = cp.require
        forbid_np[c] = cp.forbid
        tol_np[c] = cp.tolerated

    class_req = torch.from_numpy(req_np).to(dev)                    # [C,R]
    # class_tol stays HOST-side: cycle_runner dereferences it per class on
    # the CPU and passes the value into the launch (vamd_api.h)
    class_tol = torch.from_numpy(tol_np)
    class_require = torch.from_numpy(require_np).to(dev)            # [C,W]
    class_forbid = torch.from_numpy(forbid_np).to(dev)
    class_min = torch.from_numpy(
        np.ascontiguousarray(cds["min_needed"], dtype=np.int32)).to(dev)
    dim_w = plan.dim_w.to(dev, torch.float32)
    q_alloc = plan.queue_alloc.to(dev).contiguous()
    q_limit = plan.queue_limit.to(dev).contiguous()
    bias = plan.bias.to(dev) if plan.bias is not None else None
    bias_rows = torch.from_numpy(
        np.stack(bias_rows_list).astype(np.float32)).to(dev).contiguous() \
        if bias_rows_list else None                                # [B,N]

    score = torch.empty(N, dtype=torch.float32, device=dev)
    cap = torch.empty(N, dtype=torch.int32, device=dev)
    log_nodes = torch.zeros(plan.log_total, dtype=torch.int32, device=dev)
    log_counts = torch.zeros(plan.log_total, dtype=torch.int32, device=dev)
    log_len = torch.zeros(C, dtype=torch.int32, device=dev)
    class_placed = torch.zeros(C, dtype=torch.int32, device=dev)
    job_placed = torch.zeros(len(plan.jobs), dtype=torch.int32, device=dev)
    job_flag = torch.ones(len(plan.jobs), dtype=torch.uint8, device=dev)
    # radix scratch for the bulk select path (keys/ids ping-pong)
    sort_scratch = torch.empty(4 * N, dtype=torch.int32, device=dev) \
        if C and int(cds["ntasks"].max()) >= 512 and N >= 512 else None

    hip.run_cycle(
        ctypes.c_void_p(cds.ctypes.data), C,
        ctypes.c_void_p(jds.ctypes.data), len(plan.jobs),
        nt.alloc_t, nt.used_t, nt.extra_t, nt.ready, nt.taint_mask,
        nt.planes_t, bias, bias_rows, class_req, class_tol, class_require,
        class_forbid, class_min, dim_w, q_alloc, q_limit, score, cap,
        log_nodes, log_counts, log_len, class_placed, job_placed, job_flag,
        sort_scratch)

    res = CycleResult(plan, log_nodes.cpu().numpy(),
                      log_counts.cpu().numpy(), log_len.cpu().numpy(),
                      class_placed.cpu().numpy(), job_placed.cpu().numpy(),
                      job_flag.cpu().numpy())
    # the cycle mutated device queue_alloc; reflect back to the host copy
    plan.queue_alloc.copy_(q_alloc.cpu())
    return res


