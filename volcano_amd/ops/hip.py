"""ctypes bindings for the HIP kernel library + the plan-based cycle runner.

The GPU decision plane calls into ``_vamd_hip.so`` (pure HIP, C ABI).  On a
GPU box this module MUST load — ops fail loudly rather than silently
falling back to eager torch (the torch implementations in ``reference.py``
are the CPU oracle, not a GPU execution path).
"""

from __future__ import annotations

import ctypes
from ctypes import (Structure, c_float, c_int, c_int32, c_int64, c_void_p)
from typing import Optional

import torch

from .build import LIB_PATH, build

_lib = None


class VamdClassDesc(Structure):
    _fields_ = [
        ("job_idx", c_int32), ("queue_idx", c_int32), ("ntasks", c_int32),
        ("min_needed", c_int32), ("log_off", c_int32), ("log_cap", c_int32),
        ("flags", c_int32), ("bias_row", c_int32),
        ("w_least", c_float), ("w_most", c_float), ("w_bal", c_float),
        ("_padf", c_float),
    ]


class VamdJobDesc(Structure):
    _fields_ = [
        ("class_begin", c_int32), ("class_end", c_int32),
        ("occupied", c_int32), ("min_available", c_int32),
    ]


def _load():
    global _lib
    if _lib is not None:
        return _lib
    if not LIB_PATH.exists():
        try:
            build()
        except Exception as e:  # pragma: no cover
            raise RuntimeError(
                f"HIP kernel library missing and build failed: {e}. "
                "Run `python -m volcano_amd.ops.build`."
            ) from e
    lib = ctypes.CDLL(str(LIB_PATH))
    lib.vamd_score_cap.restype = None
    lib.vamd_select_commit.restype = None
    lib.vamd_finalize_job.restype = None
    lib.vamd_cond_revert.restype = None
    lib.vamd_run_cycle.restype = None
    _lib = lib
    return lib


def available() -> bool:
    try:
        _load()
        return True
    except Exception:
        return False


def _p(t: Optional[torch.Tensor]):
    return c_void_p(0 if t is None else t.data_ptr())


def _stream() -> c_void_p:
    return c_void_p(torch.cuda.current_stream().cuda_stream)


# -- per-op wrappers (used by the GPU numerics tests) -----------------------

def score_cap(alloc_t, used_t, extra_t, ready, taints, planes_t, req,
              tolerated, require, forbid, w_least, w_most, w_bal, dim_w,
              bias, score_out, cap_out):
    """All node tensors are the transposed [R, N] / [W, N] device buffers."""
    lib = _load()
    R, N = alloc_t.shape
    W = planes_t.shape[0]
    lib.vamd_score_cap(
        _p(alloc_t), _p(used_t), _p(extra_t), _p(ready), _p(taints),
        _p(planes_t), _p(req), c_int64(int(tolerated)), _p(require),
        _p(forbid), c_float(w_least), c_float(w_most), c_float(w_bal),
        _p(dim_w), _p(bias), _p(score_out), _p(cap_out),
        c_int(N), c_int(R), c_int(W), _stream())


def select_commit(score, cap, req, ntasks, used_t, queue_alloc_row,
                  queue_limit_row, log_nodes, log_counts, log_len, placed,
                  job_placed, fuse_min, sort_scratch=None):
    lib = _load()
    N = score.shape[0]
    R = req.shape[0]
    K = log_nodes.shape[0]
    lib.vamd_select_commit(
        _p(score), _p(cap), _p(req), c_int(int(ntasks)), _p(used_t),
        _p(queue_alloc_row), _p(queue_limit_row), _p(log_nodes),
        _p(log_counts), _p(log_len), _p(placed), _p(job_placed),
        c_int(int(fuse_min)), _p(sort_scratch),
        c_int(N), c_int(R), c_int(K), _stream())


def finalize_job(job_placed, occupied, min_available, class_placed,
                 class_min, flag):
    lib = _load()
    nc = class_placed.shape[0]
    lib.vamd_finalize_job(
        _p(job_placed), c_int(int(occupied)), c_int(int(min_available)),
        _p(class_placed), _p(class_min), _p(flag), c_int(nc), _stream())


def cond_revert(flag, log_nodes, log_counts, log_len, req, used_t,
                queue_alloc_row, placed, job_placed):
    lib = _load()
    R, N = used_t.shape
    lib.vamd_cond_revert(
        _p(flag), _p(log_nodes), _p(log_counts), _p(log_len), _p(req),
        _p(used_t), _p(queue_alloc_row), _p(placed), _p(job_placed),
        c_int(N), c_int(R), _stream())


# -- whole-cycle runner ------------------------------------------------------

def run_cycle(class_descs: bytes, n_classes: int, job_descs: bytes,
              n_jobs: int, alloc_t, used_t, extra_t, ready, taints, planes_t,
              bias, bias_rows, class_req, class_tol, class_require,
              class_forbid, class_min, dim_w, queue_alloc, queue_limit,
              score_scratch, cap_scratch, log_nodes, log_counts, log_len,
              class_placed, job_placed, job_flag, sort_scratch=None):
    """One library call = one whole allocate cycle (plan built host-side)."""
    lib = _load()
    R, N = alloc_t.shape
    W = planes_t.shape[0]
    lib.vamd_run_cycle(
        class_descs, c_int(n_classes),
        job_descs, c_int(n_jobs),
        _p(alloc_t), _p(used_t), _p(extra_t), _p(ready), _p(taints),
        _p(planes_t), _p(bias), _p(bias_rows), _p(class_req), _p(class_tol),
        _p(class_require), _p(class_forbid), _p(class_min), _p(dim_w),
        _p(queue_alloc), _p(queue_limit), _p(score_scratch), _p(cap_scratch),
        _p(log_nodes), _p(log_counts), _p(log_len), _p(class_placed),
        _p(job_placed), _p(job_flag), _p(sort_scratch),
        c_int(N), c_int(R), c_int(W), _stream())
