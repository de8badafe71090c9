#!/usr/bin/env python3
"""Regenerate docs/api.md from the dataclass field lists of every
registered kind (volcano_amd/api/objects.py KINDS)."""

import dataclasses
import os
import sys
import typing

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from volcano_amd.api.objects import KINDS  # noqa: E402

GROUPS = {
    "Node": "core/v1", "Pod": "core/v1",
    "PersistentVolume": "core/v1", "PersistentVolumeClaim": "core/v1",
    "PodDisruptionBudget": "policy/v1", "ResourceQuota": "core/v1",
    "Job": "batch.volcano.sh/v1alpha1", "CronJob": "batch.volcano.sh/v1alpha1",
    "PodGroup": "scheduling.volcano.sh/v1beta1",
    "Queue": "scheduling.volcano.sh/v1beta1",
    "Command": "bus.volcano.sh/v1alpha1",
    "HyperNode": "topology.volcano.sh/v1alpha1",
    "NodeShard": "shard.volcano.sh/v1alpha1",
    "JobFlow": "flow.volcano.sh/v1alpha1",
    "JobTemplate": "flow.volcano.sh/v1alpha1",
    "Numatopology": "nodeinfo.volcano.sh/v1alpha1",
    "ColocationConfig": "config.volcano.sh/v1alpha1",
    "HyperJob": "batch.volcano.sh/v1alpha1",
}


def type_name(t) -> str:
    s = str(t).replace("typing.", "")
    for tok in ("volcano_amd.api.objects.", "volcano_amd.api.resource.",
                "<class '", "'>"):
        s = s.replace(tok, "")
    return s


def main() -> None:
    out = ["# API reference (CRD-shaped kinds)", "",
           "Generated from `volcano_amd/api/objects.py` "
           "(`python tools/gen_api_docs.py`). All kinds round-trip",
           "through `to_dict`/`from_dict` (JSON/YAML), the object store, "
           "the", "REST apiserver (`/apis/{kind}`), and `vcctl -f` "
           "where applicable.", ""]
    for kind in sorted(KINDS):
        cls = KINDS[kind]
        out.append(f"## {kind}  — `{GROUPS.get(kind, 'core/v1')}`")
        out.append("")
        doc = (cls.__doc__ or "").strip().splitlines()
        if doc:
            out.append(doc[0].rstrip("."))
        out.append("")
        if dataclasses.is_dataclass(cls):
            for f in dataclasses.fields(cls):
                if f.name.startswith("_"):
                    continue
                out.append(f"- `{f.name}`: {type_name(f.type)}")
        out.append("")
    path = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "docs", "api.md")
    with open(path, "w") as fh:
        fh.write("\n".join(out))
    print(f"wrote {path} ({len(KINDS)} kinds)")


if __name__ == "__main__":
    main()
