#!/usr/bin/env python3
"""Load a queue hierarchy YAML into the state file (see queues.yaml)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import yaml

from volcano_amd.api.objects import ObjectMeta, Queue, QueueSpec
from volcano_amd.api.resource import Resource
from volcano_amd.store import ObjectStore


def main():
    conf = sys.argv[1] if len(sys.argv) > 1 else "examples/queues.yaml"
    state = sys.argv[2] if len(sys.argv) > 2 else "/tmp/volcano-amd-state.json"
    try:
        store = ObjectStore.load(state)
    except FileNotFoundError:
        store = ObjectStore()
    with open(conf) as f:
        data = yaml.safe_load(f)
    for q in data.get("queues", []):
        queue = Queue(
            meta=ObjectMeta(name=q["name"]),
            spec=QueueSpec(
                weight=int(q.get("weight", 1)),
                parent=q.get("parent", ""),
                reclaimable=bool(q.get("reclaimable", True)),
                deserved=Resource.from_spec(q.get("deserved")),
                guarantee=Resource.from_spec(q.get("guarantee")),
                capability=Resource.from_spec(q.get("capability"))))
        store.apply("Queue", queue)
        print(f"queue {q['name']} applied")
    store.save(state)


if __name__ == "__main__":
    main()
