"""Extender plugin (reference ``plugins/extender``: the HTTP sidecar
protocol — POSTs JSON to an external scheduler extender at nearly every
extension point, extender.go:354-390).

Arguments: {"extender.urlPrefix": "http://host:port",
            "extender.predicateVerb": "predicate",
            "extender.jobEnqueueableVerb": "jobEnqueueable",
            "extender.preemptableVerb": "preemptable",
            "extender.onSessionOpenVerb": "...",
            "extender.onSessionCloseVerb": "...",
            "extender.ignorable": true}

MI355X batching departure: the reference POSTs per (task, node) batch
per task; here ONE call per task class carries every node name and the
response's infeasible set becomes a dynamic forbid bit — the sidecar
round-trip leaves the kernel hot loop untouched.
"""

from __future__ import annotations

import json
from typing import Dict, Optional

from ..session import ABSTAIN, PERMIT, REJECT
from ..tensors import set_plane_bit
from .base import Plugin, register


@register("extender")
class ExtenderPlugin(Plugin):
    def _post(self, verb: str, payload: dict) -> Optional[dict]:
        if not verb:
            return None
        import httpx
        url = f"{self.prefix}/{verb}"
        try:
            r = httpx.post(url, json=payload, timeout=self.timeout)
            r.raise_for_status()
            return r.json()
        except Exception:
            if self.ignorable:
                return None
            raise

    def on_session_open(self, ssn) -> None:
        a = self.args
        self.prefix = str(a.get("extender.urlPrefix", "")).rstrip("/")
        if not self.prefix:
            return
        self.timeout = float(a.get("extender.httpTimeout", 5.0))
        self.ignorable = bool(a.get("extender.ignorable", True))
        nt = ssn.node_tensors

        self._post(a.get("extender.onSessionOpenVerb"), {
            "jobs": list(ssn.jobs),
            "nodes": list(ssn.nodes),
            "queues": list(ssn.queues),
        })

        predicate_verb = a.get("extender.predicateVerb")
        if predicate_verb:
            memo: Dict[str, Optional[int]] = {}

            def hook(tclass, job, require, forbid):
                bit = memo.get(tclass.signature)
                if bit is None and tclass.signature not in memo:
                    t = tclass.tasks[0]
                    resp = self._post(predicate_verb, {
                        "task": {"name": t.name, "namespace": t.namespace,
                                 "job": t.job_key,
                                 "resreq": dict(t.request.q)},
                        "nodes": list(ssn.nodes),
                    })
                    bit = None
                    if resp is not None:
                        feasible = set(resp.get("nodes",
                                                resp.get("feasibleNodes", [])))
                        unfit = [ni.node_id for name, ni in ssn.nodes.items()
                                 if name not in feasible]
                        if unfit:
                            bit = nt.add_dynamic_bit(
                                f"ext:{tclass.signature}", unfit)
                    memo[tclass.signature] = bit
                if bit is not None:
                    set_plane_bit(forbid, bit)

            ssn.class_constraint_hooks.append(hook)

        enq_verb = a.get("extender.jobEnqueueableVerb")
        if enq_verb:
            def job_enqueueable(job) -> int:
                resp = self._post(enq_verb, {"job": job.key,
                                             "queue": job.queue})
                if resp is None:
                    return ABSTAIN
                v = resp.get("status", resp.get("result"))
                if v in (True, "permit", 1):
                    return PERMIT
                if v in (False, "reject", -1):
                    return REJECT
                return ABSTAIN
            ssn.job_enqueueable_fns.append(job_enqueueable)

        pre_verb = a.get("extender.preemptableVerb")
        if pre_verb:
            def preemptable(preemptor, candidates):
                resp = self._post(pre_verb, {
                    "preemptor": preemptor.key,
                    "candidates": [v.key for v in candidates]})
                if resp is None:
                    return candidates
                keep = set(resp.get("victims", []))
                return [v for v in candidates if v.key in keep]
            ssn.preemptable_fns.append(preemptable)

    def on_session_close(self, ssn) -> None:
        if getattr(self, "prefix", ""):
            self._post(self.args.get("extender.onSessionCloseVerb"), {})
