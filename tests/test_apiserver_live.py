"""Live multi-process control plane: uvicorn apiserver + HTTP client +
vcctl --server + scheduler against the same store."""

import socket
import time

import pytest

from volcano_amd.scheduler import FakeBinder, Scheduler, SchedulerCache
from volcano_amd.store import ObjectStore
from volcano_amd.store.apiserver import serve
from volcano_amd.store.client import StoreClient
from volcano_amd.utils import synth

GI = 1024 ** 3


@pytest.fixture(scope="module")
def live():
    store = ObjectStore()
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    server = serve(store, port=port)
    client = StoreClient(f"http://127.0.0.1:{port}")
    for _ in range(100):
        if client.healthz():
            break
        time.sleep(0.05)
    else:
        pytest.fail("apiserver did not come up")
    yield store, client, port
    server.should_exit = True


def test_http_crud_watch_and_schedule(live):
    store, client, port = live
    for n in synth.make_nodes(2, cpu_milli=4000, mem=16 * GI):
        client.create("Node", n)
    client.create("Queue", synth.make_queue("default"))
    rv0 = store.resource_version

    pg = synth.make_podgroup("hj", min_member=2)
    client.create("PodGroup", pg)
    for i in range(2):
        client.create("Pod", synth.make_pod(f"hj-w-{i}", "hj",
                                            cpu_milli=1000, mem=GI))

    # watch over HTTP sees the writes
    rv, events = client.watch_since(rv0, ("PodGroup", "Pod"))
    assert len(events) == 3

    # scheduler runs against the same (server-side) store
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    Scheduler(cache).run_once()
    assert len(binder.binds) == 2

    # admission over HTTP rejects bad objects
    from volcano_amd.api.objects import Job, JobSpec, ObjectMeta, TaskSpec
    bad = Job(meta=ObjectMeta(name="bad"),
              spec=JobSpec(min_available=99, tasks=[
                  TaskSpec(name="w", replicas=1, template={})]))
    from volcano_amd.store.client import ApiError
    with pytest.raises(ApiError):
        client.create("Job", bad)


def test_vcctl_against_live_server(live, capsys):
    store, client, port = live
    from volcano_amd.cli.vcctl import main
    url = f"http://127.0.0.1:{port}"
    assert main(["--server", url, "queue", "create", "-N", "web", "-w", "2"]) == 0
    assert main(["--server", url, "job", "run", "-N", "served", "-r", "2",
                 "-q", "web"]) == 0
    assert main(["--server", url, "job", "list"]) == 0
    out = capsys.readouterr().out
    assert "served" in out
    assert store.get("Job", "default", "served") is not None
    assert main(["--server", url, "job", "delete", "-N", "served"]) == 0
    assert store.get("Job", "default", "served") is None
