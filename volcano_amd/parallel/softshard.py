"""Soft sharding — replicated nodes, sharded jobs, RCCL delta exchange.

The second coordination mode of SURVEY §2.9 C2 (the first, hard node
sharding, is `sharding.py`): every rank sees ALL nodes (so a gang may
span the whole cluster), jobs are sharded by hash, and each cycle the
ranks reconcile their staged placements by exchanging the per-node usage
DELTA tensors over RCCL/xGMI:

1. every rank runs its plan against its local copy of ``used``;
2. all_gather of ``delta = used_after − used_before`` ([R, N] f32 —
   ~320 KB at 10k nodes, latency-bound on xGMI);
3. deterministic admission: rank order prefix — rank r's placements on
   node n stand iff the cumulative delta of ranks 0..r fits the node's
   allocatable in every dimension;
4. a rank reverts EVERY job that placed on a node it lost (gang
   atomicity; `AllocateAction._apply(bad_nodes=...)`), which only frees
   capacity, so the admitted prefix stays feasible;
5. all_reduce(SUM) of the post-revert deltas establishes the global
   ``used``; each rank folds other ranks' usage into its host mirror as
   per-node ``remote_used``.

Losing jobs return to Pending and re-place next cycle against the now
visible global usage.
"""

from __future__ import annotations

from typing import FrozenSet

import torch


class SoftShardCoordinator:
    # conflict-loser repulsion: a node this rank LOST in the reconcile
    # gets a transient extra penalty (decaying ×0.5/cycle) so adversarial
    # same-score workloads stop re-colliding on the same nodes cycle
    # after cycle.  Amplitude stays tie-break scale — real score
    # differences (and feasibility) always dominate.
    REPEL = 1e-4
    REPEL_DECAY = 0.5
    REPEL_FLOOR = 1e-7

    def __init__(self, rank: int, world: int):
        self.rank = rank
        self.world = world
        self._stagger = None
        self._repel = {}               # node_id -> penalty strength
        # measured-mode stats (bench JSON): reconcile rounds and nodes
        # this rank lost in the cross-rank admission
        self.reconciles = 0
        self.conflict_nodes = 0

    def stagger_bias(self, nt) -> torch.Tensor:
        """Rank-staggered tie-break: an epsilon-scale score bias that makes
        each rank prefer a different region of the node space, so equal-
        score placements from different ranks don't collide on the same
        low-index nodes (real score differences still dominate)."""
        if self._stagger is None or self._stagger.shape[0] != nt.n:
            N = nt.n
            offset = (self.rank * N) // max(self.world, 1)
            order = (torch.arange(N, dtype=torch.float32) - offset) % max(N, 1)
            self._stagger = (-1e-6 * order).to(nt.alloc_t.device)
        if not self._repel:
            return self._stagger
        bias = self._stagger.clone()
        ids = torch.tensor(list(self._repel), dtype=torch.long,
                           device=bias.device)
        pen = torch.tensor(list(self._repel.values()), dtype=torch.float32,
                           device=bias.device)
        bias.index_add_(0, ids, -pen)
        return bias

    def _note_losses(self, lost_ids) -> None:
        """Decay old repulsion, strengthen freshly-lost nodes."""
        decayed = {}
        for nid, s in self._repel.items():
            s *= self.REPEL_DECAY
            if s > self.REPEL_FLOOR:
                decayed[nid] = s
        for nid in lost_ids:
            decayed[nid] = decayed.get(nid, 0.0) + self.REPEL
        self._repel = decayed

    # -- step 2+3: gather deltas, find this rank's lost nodes ----------------
    @staticmethod
    def _comm_dev(t: torch.Tensor) -> torch.Tensor:
        """Collectives run on the backend's native device: RCCL moves
        device tensors over xGMI; gloo needs host staging."""
        if torch.distributed.get_backend() == "gloo" and t.is_cuda:
            return t.cpu()
        return t.contiguous()

    def find_conflicts(self, nt, used_before: torch.Tensor) -> FrozenSet[int]:
        delta = nt.used_t - used_before
        if self.world <= 1 or not torch.distributed.is_initialized():
            return frozenset()
        comm = self._comm_dev(delta)
        gathered = [torch.zeros_like(comm) for _ in range(self.world)]
        torch.distributed.all_gather(gathered, comm)
        gathered = [g.to(delta.device) for g in gathered]
        stack = torch.stack(gathered)                    # [world, R, N]
        cum = torch.cumsum(stack, dim=0)
        base = used_before.unsqueeze(0)                  # staged-before state
        ok = ((base + cum) <= nt.alloc_t.unsqueeze(0) + 0.1).all(dim=1)  # [world, N]
        mine_active = delta.abs().sum(dim=0) > 1e-6      # [N]
        lost = (~ok[self.rank]) & mine_active
        self._gathered_sum = stack.sum(dim=0)
        ids = torch.nonzero(lost, as_tuple=False).flatten()
        out = frozenset(int(i) for i in ids.cpu())
        self.reconciles += 1
        self.conflict_nodes += len(out)
        self._note_losses(out)
        return out

    # -- step 5: establish the global used state -----------------------------
    def finalize(self, ssn, nt, used_before: torch.Tensor) -> None:
        if self.world <= 1 or not torch.distributed.is_initialized():
            return
        my_delta = (nt.used_t - used_before).contiguous()
        total = self._comm_dev(my_delta).clone()
        torch.distributed.all_reduce(total, op=torch.distributed.ReduceOp.SUM)
        total = total.to(my_delta.device)
        remote = total - my_delta                        # other ranks' usage
        nt.used_t.copy_(used_before + total)
        # fold remote usage into the host mirror so future packs and the
        # preempt path see the global state
        remote_nr = remote.t().cpu().numpy()             # [N, R]
        import numpy as np
        ledger = getattr(ssn.cache, "ledger", None)
        if ledger is not None and ledger.n == remote_nr.shape[0]:
            # vectorized fold into the columnar accounting (both the used
            # truth and the remote-attribution plane)
            from ..api.ledger import REMOTE, USED
            R = remote_nr.shape[1]
            if ledger.width < R:
                ledger._widen(R)
            ledger.planes[USED, :, :R] += remote_nr
            ledger.planes[REMOTE, :, :R] += remote_nr
            ledger.version += 1
        else:
            from ..api.resource import Resource
            names = nt.dims.names
            nodes_sorted = getattr(ssn.cache, "nodes_sorted", None)
            if nodes_sorted is None or len(nodes_sorted) != len(ssn.nodes):
                nodes_sorted = sorted(ssn.nodes.values(),
                                      key=lambda n: n.name)
            hot = np.nonzero(np.abs(remote_nr).sum(axis=1) > 1e-6)[0]
            for i in hot:
                ni = nodes_sorted[int(i)]
                delta = Resource({names[r]: float(remote_nr[i, r])
                                  for r in range(len(names))
                                  if remote_nr[i, r]})
                ni._acct(delta, 1, 0, 0)
                ni._remote_used.add(delta)
