"""Scheduler daemon entrypoint (the ``cmd/scheduler`` analog).

Run: ``python -m volcano_amd.scheduler.daemon --state /path/state.json``
or with ``--serve-api`` to also expose the apiserver for vcctl.

Signals (reference cache/dumper.go + SIGUSR1/2): SIGUSR1 dumps the
scheduler cache as JSON to --dump-dir.
"""

from __future__ import annotations

import argparse
import json
import signal
import sys
import time


def dump_cache(cache, path: str) -> None:
    """Cache dumper (reference pkg/scheduler/cache/dumper.go:127-146)."""
    state = {
        "nodes": {
            name: {
                "allocatable": dict(ni.allocatable.q),
                "used": dict(ni.used.q),
                "releasing": dict(ni.releasing.q),
                "tasks": sorted(ni.tasks),
            } for name, ni in cache.nodes.items()
        },
        "jobs": {
            key: {
                "queue": job.queue,
                "min_available": job.min_available,
                "phase": job.phase,
                "tasks": {t.key: {"status": t.status.name,
                                  "node": t.node_name}
                          for t in job.tasks.values()},
            } for key, job in cache.jobs.items()
        },
        "queues": sorted(cache.queues),
        "ts": time.time(),
    }
    with open(path, "w") as f:
        json.dump(state, f, indent=1)


def main(argv=None) -> int:
    from ..store import ObjectStore
    from ..utils.metrics import METRICS
    from .cache import SchedulerCache
    from .config import SchedulerConfiguration, default_config
    from .engine import Scheduler

    ap = argparse.ArgumentParser(prog="volcano-amd-scheduler")
    ap.add_argument("--state", default="/tmp/volcano-amd-state.json")
    ap.add_argument("--conf", default=None,
                    help="scheduler YAML conf (actions + tiers)")
    ap.add_argument("--period", type=float, default=1.0)
    ap.add_argument("--device", default="auto",
                    help="cuda|cpu|auto for the decision plane")
    ap.add_argument("--serve-api", action="store_true")
    ap.add_argument("--api-port", type=int, default=8343)
    ap.add_argument("--dump-dir", default="/tmp")
    ap.add_argument("--once", action="store_true")
    ap.add_argument("--plugins-dir", default=None,
                    help="directory of out-of-tree plugin modules "
                         "(the reference's --plugins-dir)")
    args = ap.parse_args(argv)

    import torch
    device = args.device
    if device == "auto":
        device = "cuda" if torch.cuda.is_available() else "cpu"

    # long-lived daemon with a large object graph: raise the cyclic-GC
    # thresholds so gen-2 scans don't land inside scheduling cycles
    # (docs/tuning.md; the bench freezes its permanent inventory outright)
    import gc
    gc.set_threshold(100000, 50, 50)

    try:
        store = ObjectStore.load(args.state)
    except FileNotFoundError:
        store = ObjectStore()

    if args.plugins_dir:
        from .plugins import load_plugins_dir
        loaded = load_plugins_dir(args.plugins_dir)
        if loaded:
            print(f"loaded out-of-tree plugins: {', '.join(loaded)}",
                  file=sys.stderr, flush=True)

    if args.conf:
        with open(args.conf) as f:
            config = SchedulerConfiguration.from_yaml(f.read())
    else:
        config = default_config()
    config.use_hip = device == "cuda"
    config.device = device
    config.schedule_period = args.period

    cache = SchedulerCache(store=store, device=device)
    sched = Scheduler(cache, config)

    if args.serve_api:
        from ..store.apiserver import serve
        serve(store, port=args.api_port)
        print(f"apiserver on :{args.api_port}", flush=True)

    def on_usr1(sig, frame):
        path = f"{args.dump_dir}/volcano-amd-cache-{int(time.time())}.json"
        dump_cache(cache, path)
        print(f"cache dumped to {path}", file=sys.stderr, flush=True)

    signal.signal(signal.SIGUSR1, on_usr1)

    # SIGUSR2: the --enable-pprof analog (reference cmd/scheduler/app/
    # server.go:162) — profile the NEXT cycle with cProfile and write
    # pstats + a text summary next to the cache dumps
    def on_usr2(sig, frame):
        import cProfile
        import io
        import pstats
        stamp = int(time.time())
        orig = sched.run_once

        def profiled_once():
            sched.run_once = orig       # one-shot
            pr = cProfile.Profile()
            pr.enable()
            out = orig()
            pr.disable()
            base = f"{args.dump_dir}/volcano-amd-profile-{stamp}"
            pr.dump_stats(base + ".pstats")
            buf = io.StringIO()
            pstats.Stats(pr, stream=buf).sort_stats(
                "cumulative").print_stats(40)
            with open(base + ".txt", "w") as f:
                f.write(buf.getvalue())
            print(f"cycle profile written to {base}.txt",
                  file=sys.stderr, flush=True)
            return out

        sched.run_once = profiled_once

    signal.signal(signal.SIGUSR2, on_usr2)

    if args.once:
        sched.run_once()
        store.save(args.state)
        print(METRICS.export_text())
        return 0

    # conf hot-reload (reference scheduler.go:165-215 fsnotify watcher —
    # here an mtime poll folded into the cycle loop)
    if args.conf:
        import os
        conf_mtime = os.stat(args.conf).st_mtime
        orig_run_once = sched.run_once

        def run_once_with_reload():
            nonlocal conf_mtime
            try:
                m = os.stat(args.conf).st_mtime
                if m != conf_mtime:
                    conf_mtime = m
                    with open(args.conf) as f:
                        new_conf = SchedulerConfiguration.from_yaml(f.read())
                    new_conf.use_hip = config.use_hip
                    new_conf.device = config.device
                    sched.config = new_conf
                    from . import actions as actions_mod
                    sched._actions = [actions_mod.new_action(a)
                                      for a in new_conf.actions]
                    print("scheduler conf reloaded", file=sys.stderr,
                          flush=True)
            except FileNotFoundError:
                pass
            return orig_run_once()

        sched.run_once = run_once_with_reload
    try:
        # long-lived inventory (node/pod mirrors, tensors) out of the
        # cyclic collector's gen-2 scan set: avoids multi-hundred-ms GC
        # stalls under sustained churn (young garbage still collects)
        import gc
        gc.collect()
        gc.freeze()
        sched.run(period=args.period)
    except KeyboardInterrupt:
        pass
    store.save(args.state)
    return 0


if __name__ == "__main__":
    sys.exit(main())
