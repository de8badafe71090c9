"""All-in-one launcher — the installer/helm-chart analog: one process
hosting the apiserver (REST + admission), the controller manager, the
scheduler loop and optional node agents, sharing one store.

  python -m volcano_amd.launcher --api-port 8343 --state /tmp/cluster.json

Then from anywhere:
  vcctl --server http://127.0.0.1:8343 queue create -N prod
  vcctl --server http://127.0.0.1:8343 job run -N train -r 8 -q prod
"""

from __future__ import annotations

import argparse
import signal
import sys
import time


def main(argv=None) -> int:
    from .controllers.framework import ControllerManager
    from .scheduler.cache import SchedulerCache
    from .scheduler.config import SchedulerConfiguration, default_config
    from .scheduler.engine import Scheduler
    from .store import ObjectStore
    from .store.apiserver import serve

    ap = argparse.ArgumentParser(prog="volcano-amd")
    ap.add_argument("--state", default="/tmp/volcano-amd-state.json")
    ap.add_argument("--api-port", type=int, default=8343)
    ap.add_argument("--period", type=float, default=0.5)
    ap.add_argument("--conf", default=None)
    ap.add_argument("--device", default="auto")
    ap.add_argument("--agents", action="store_true",
                    help="run node agents (oversubscription/QoS/eviction) "
                         "for every node in the store")
    args = ap.parse_args(argv)

    import torch
    device = args.device
    if device == "auto":
        device = "cuda" if torch.cuda.is_available() else "cpu"

    try:
        store = ObjectStore.load(args.state)
    except FileNotFoundError:
        store = ObjectStore()

    serve(store, port=args.api_port)
    cm = ControllerManager(store, period=0.1)
    cm.run()

    if args.conf:
        with open(args.conf) as f:
            config = SchedulerConfiguration.from_yaml(f.read())
    else:
        config = default_config()
    config.use_hip = device == "cuda"
    config.device = device
    cache = SchedulerCache(store=store, device=device)
    sched = Scheduler(cache, config)

    agent_stop = None
    if args.agents:
        import threading

        from .agent import (CpuQosHandler, EventsManager, EvictionHandler,
                            MemoryQosHandler, OversubscriptionHandler)

        agent_stop = threading.Event()

        def agents_loop():
            managers = {}
            while not agent_stop.is_set():
                for node in store.list("Node"):
                    name = node.meta.name
                    if name not in managers:
                        m = EventsManager(store, name)
                        m.register(OversubscriptionHandler())
                        m.register(EvictionHandler())
                        m.register(CpuQosHandler())
                        m.register(MemoryQosHandler())
                        managers[name] = m
                for m in managers.values():
                    m.tick()
                agent_stop.wait(5.0)

        threading.Thread(target=agents_loop, daemon=True).start()

    stopping = []

    def on_term(sig, frame):
        stopping.append(True)
        sched.stop()

    signal.signal(signal.SIGTERM, on_term)
    signal.signal(signal.SIGINT, on_term)
    print(f"volcano-amd up: api :{args.api_port}, scheduler on {device}, "
          f"{len(cm.controllers)} controllers", flush=True)
    try:
        sched.run(period=args.period)
    finally:
        if agent_stop is not None:
            agent_stop.set()
        cm.stop()
        store.save(args.state)
        print("state saved", flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
