"""volcano_amd — an MI355X-native batch / gang scheduling framework.

A from-scratch rebuild of the capabilities of volcano-sh/volcano (the CNCF
Kubernetes batch scheduler) designed GPU-first for AMD Instinct MI355X
(gfx950 / CDNA4):

* the *decision plane* (the scheduler hot loop: per-cycle Session snapshot →
  predicate filter → score → gang commit → preempt/reclaim) runs as dense
  tensor operations in HBM3E, with the hot per-node kernels hand-written in
  HIP for CDNA4 (see ``volcano_amd/ops/csrc``) and dispatched through
  PyTorch-ROCm;
* PodGroups are sharded across the GPUs of one node; the ranks exchange
  per-node allocation deltas each cycle with RCCL collectives over xGMI
  (``volcano_amd/parallel``);
* the *control plane* (CRD-style API objects, job/queue/jobflow controllers,
  admission webhooks, the ``vcctl`` CLI and an object store standing in for
  kube-apiserver) is host-side Python (the reference's is Go/client-go;
  this environment has no Go toolchain and no cluster, so the control plane
  is a faithful re-design rather than a port — see README.md).

Reference layer map: /root/repo/SURVEY.md (volcano-sh/volcano @ 2026-08-21).
"""

from .version import __version__  # noqa: F401
