"""Controller-manager entrypoint (the ``cmd/controller-manager`` analog).

Run: ``python -m volcano_amd.controllers.daemon --state state.json``.
"""

from __future__ import annotations

import argparse
import sys
import time


def main(argv=None) -> int:
    from ..store import ObjectStore
    from .framework import CONTROLLER_REGISTRY, ControllerManager

    ap = argparse.ArgumentParser(prog="volcano-amd-controller-manager")
    ap.add_argument("--state", default="/tmp/volcano-amd-state.json")
    ap.add_argument("--controllers", default="*",
                    help=f"comma list or * (known: {sorted(CONTROLLER_REGISTRY)})")
    ap.add_argument("--period", type=float, default=0.2)
    ap.add_argument("--once", action="store_true")
    args = ap.parse_args(argv)

    try:
        store = ObjectStore.load(args.state)
    except FileNotFoundError:
        store = ObjectStore()

    names = None if args.controllers == "*" else \
        [c.strip() for c in args.controllers.split(",") if c.strip()]
    cm = ControllerManager(store, names, period=args.period)
    if args.once:
        cm.sync_until_quiet()
        store.save(args.state)
        return 0
    try:
        cm.run()
        while True:
            time.sleep(5)
            store.save(args.state)
    except KeyboardInterrupt:
        cm.stop()
        store.save(args.state)
    return 0


if __name__ == "__main__":
    sys.exit(main())
