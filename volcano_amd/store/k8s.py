"""Kubernetes wire-shape translation: group/version/kind mapping and
manifest conversion between k8s-style documents (apiVersion/kind/
metadata/spec, camelCase keys) and the internal dataclasses (meta/spec,
snake_case) — so standard tooling and manifests written against the
reference's CRDs (/root/reference/staging/src/volcano.sh/apis/) apply
unchanged through the HTTP server's ``/apis/{group}/{version}/...``
paths.
"""

from __future__ import annotations

import re
from typing import Any, Dict, Optional, Tuple

from ..api.objects import KINDS, from_dict, to_dict

# kind → (group, version, plural).  "" group = core (/api/v1).
GVK: Dict[str, Tuple[str, str, str]] = {
    "Job": ("batch.volcano.sh", "v1alpha1", "jobs"),
    "CronJob": ("batch.volcano.sh", "v1alpha1", "cronjobs"),
    "PodGroup": ("scheduling.volcano.sh", "v1beta1", "podgroups"),
    "Queue": ("scheduling.volcano.sh", "v1beta1", "queues"),
    "Command": ("bus.volcano.sh", "v1alpha1", "commands"),
    "HyperNode": ("topology.volcano.sh", "v1alpha1", "hypernodes"),
    "NodeShard": ("shard.volcano.sh", "v1alpha1", "nodeshards"),
    "JobFlow": ("flow.volcano.sh", "v1alpha1", "jobflows"),
    "JobTemplate": ("flow.volcano.sh", "v1alpha1", "jobtemplates"),
    "Numatopology": ("nodeinfo.volcano.sh", "v1alpha1", "numatopologies"),
    "ColocationConfig": ("config.volcano.sh", "v1alpha1",
                         "colocationconfigs"),
    "HyperJob": ("training.volcano.sh", "v1alpha1", "hyperjobs"),
    "Pod": ("", "v1", "pods"),
    "Namespace": ("", "v1", "namespaces"),
    "Node": ("", "v1", "nodes"),
    "PersistentVolume": ("", "v1", "persistentvolumes"),
    "PersistentVolumeClaim": ("", "v1", "persistentvolumeclaims"),
    "ResourceQuota": ("", "v1", "resourcequotas"),
    "PodDisruptionBudget": ("policy", "v1", "poddisruptionbudgets"),
    "DeviceClass": ("resource.k8s.io", "v1", "deviceclasses"),
    "ResourceClaim": ("resource.k8s.io", "v1", "resourceclaims"),
    "DataSource": ("datadependency.volcano.sh", "v1alpha1", "datasources"),
    "DataSourceClaim": ("datadependency.volcano.sh", "v1alpha1",
                        "datasourceclaims"),
}

# (group, version, plural) → kind
_BY_PATH: Dict[Tuple[str, str, str], str] = {
    (g, v, p): k for k, (g, v, p) in GVK.items()}

# keys whose VALUES are opaque user maps (label keys, resource names,
# plugin arguments ...) — key-case conversion must not descend into them
_OPAQUE = frozenset({
    "labels", "annotations", "nodeSelector", "node_selector",
    "matchLabels", "match_labels", "labelMatch", "label_match",
    "template", "resources", "arguments", "selector", "capability",
    "guarantee", "deserved", "minResources", "min_resources",
    "allocatable", "oversubscription", "request", "hard", "patch",
    "networkTopology", "network_topology", "affinity", "extra",
    "minTaskMember", "min_task_member", "plugins", "settings",
})

_CAMEL_RE = re.compile(r"_([a-z0-9])")
_SNAKE_RE = re.compile(r"(?<!^)(?=[A-Z])")


def _to_camel(key: str) -> str:
    return _CAMEL_RE.sub(lambda m: m.group(1).upper(), key)


def _to_snake(key: str) -> str:
    return _SNAKE_RE.sub("_", key).lower()


def _convert(data: Any, fn) -> Any:
    if isinstance(data, dict):
        out = {}
        for k, v in data.items():
            nk = fn(k) if isinstance(k, str) else k
            out[nk] = v if (k in _OPAQUE or nk in _OPAQUE) \
                else _convert(v, fn)
        return out
    if isinstance(data, list):
        return [_convert(v, fn) for v in data]
    return data


def api_version(kind: str) -> str:
    g, v, _ = GVK[kind]
    return f"{g}/{v}" if g else v


def kind_for(group: str, version: str, plural: str) -> Optional[str]:
    return _BY_PATH.get((group, version, plural))


def to_manifest(obj) -> Dict[str, Any]:
    """Internal object → k8s-style manifest (camelCase, metadata:)."""
    kind = type(obj).__name__
    d = to_dict(obj)
    meta = d.pop("meta", {})
    body = _convert(d, _to_camel)
    out = {"apiVersion": api_version(kind), "kind": kind,
           "metadata": _convert(meta, _to_camel)}
    out.update(body)
    return out


def from_manifest(data: Dict[str, Any], kind: Optional[str] = None):
    """k8s-style manifest → internal object.  Accepts both the k8s shape
    (metadata/camelCase) and the internal shape (meta/snake_case)."""
    kind = kind or data.get("kind")
    cls = KINDS.get(kind or "")
    if cls is None:
        raise KeyError(f"unknown kind {kind!r}")
    d = dict(data)
    d.pop("apiVersion", None)
    d.pop("kind", None)
    if "metadata" in d and "meta" not in d:
        d["meta"] = d.pop("metadata")
    d = _convert(d, _to_snake)
    # tolerate k8s metadata fields the internal ObjectMeta doesn't carry
    import dataclasses
    from ..api.objects import ObjectMeta
    meta_fields = {f.name for f in dataclasses.fields(ObjectMeta)}
    if isinstance(d.get("meta"), dict):
        d["meta"] = {k: v for k, v in d["meta"].items() if k in meta_fields}
    top_fields = {f.name for f in dataclasses.fields(cls)}
    d = {k: v for k, v in d.items() if k in top_fields}
    return from_dict(cls, d)


def crd_manifest(kind: str) -> Optional[Dict[str, Any]]:
    """CustomResourceDefinition manifest for one of the volcano API
    groups (None for core kinds).  Structural schema is permissive
    (x-kubernetes-preserve-unknown-fields) — the authoritative
    validation lives in the admission chain, as in the reference's
    webhook-manager."""
    g, v, plural = GVK[kind]
    if not g or g in ("policy", "resource.k8s.io"):
        return None
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {"name": f"{plural}.{g}"},
        "spec": {
            "group": g,
            "names": {
                "kind": kind,
                "listKind": f"{kind}List",
                "plural": plural,
                "singular": kind.lower(),
            },
            "scope": "Namespaced" if kind not in (
                "Queue", "HyperNode", "NodeShard") else "Cluster",
            "versions": [{
                "name": v,
                "served": True,
                "storage": True,
                "schema": {"openAPIV3Schema": {
                    "type": "object",
                    "properties": {
                        "spec": {"type": "object",
                                 "x-kubernetes-preserve-unknown-fields": True},
                        "status": {"type": "object",
                                   "x-kubernetes-preserve-unknown-fields": True},
                    },
                    "x-kubernetes-preserve-unknown-fields": True,
                }},
                "subresources": {"status": {}},
            }],
        },
    }
