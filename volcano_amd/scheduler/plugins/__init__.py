"""Policy plugins (reference ``pkg/scheduler/plugins/``).

Each plugin registers host-side ordering/vote callbacks and/or tensor-plane
configuration on the Session at ``on_session_open``.  Registry pattern
mirrors the reference's ``plugins/factory.go``.
"""

from .base import PLUGIN_REGISTRY, Plugin, register
from . import (binpack, capacity, cdp, conformance, deviceshare, drf,
               extender, gang, imagelocality, interpodaffinity,
               network_topology_aware, nodegroup, nodeorder, numaaware,
               overcommit, pdb, predicates, priority, proportion,
               rescheduling, resource_strategy_fit, resourcequota, sla,
               task_topology, tdm, topologyspread,
               usage)  # noqa: F401 (side-effect registration)


def new_plugin(name: str, args=None) -> Plugin:
    try:
        factory = PLUGIN_REGISTRY[name]
    except KeyError:
        raise KeyError(f"unknown plugin {name!r}; known: {sorted(PLUGIN_REGISTRY)}")
    return factory(args or {})
