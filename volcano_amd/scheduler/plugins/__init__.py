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


def load_plugins_dir(path: str) -> list:
    """Load out-of-tree plugin modules from a directory (the reference's
    ``--plugins-dir`` dlopen mechanism, framework/plugins.go): every
    ``*.py`` file is imported; modules self-register via the
    ``@register(name)`` decorator and are then addressable from the
    scheduler conf tiers like any built-in.  Returns the names newly
    registered."""
    import importlib.util
    import os
    before = set(PLUGIN_REGISTRY)
    if not path or not os.path.isdir(path):
        return []
    for fn in sorted(os.listdir(path)):
        if not fn.endswith(".py") or fn.startswith("_"):
            continue
        mod_name = f"volcano_amd_ext_{fn[:-3]}"
        spec = importlib.util.spec_from_file_location(
            mod_name, os.path.join(path, fn))
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
    return sorted(set(PLUGIN_REGISTRY) - before)
