"""Daemon entrypoints (--once), metrics exposition."""

import json

from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth
from volcano_amd.utils.metrics import METRICS


def test_scheduler_and_controller_daemons_once(tmp_path, capsys):
    from volcano_amd.api.objects import Job, JobSpec, ObjectMeta, TaskSpec
    from volcano_amd.controllers.daemon import main as cm_main
    from volcano_amd.scheduler.daemon import main as sched_main

    state = str(tmp_path / "state.json")
    store = ObjectStore()
    for n in synth.make_nodes(2, cpu_milli=4000, mem=8 * 1024 ** 3):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    store.create("Job", Job(meta=ObjectMeta(name="d1"), spec=JobSpec(
        tasks=[TaskSpec(name="w", replicas=2,
                        template={"resources": {"cpu": "1",
                                                "memory": "1Gi"}})])))
    store.save(state)

    assert cm_main(["--state", state, "--once",
                    "--controllers", "job,podgroup,queue"]) == 0
    assert sched_main(["--state", state, "--once"]) == 0
    out = capsys.readouterr().out
    assert "e2e_scheduling_latency" in out     # metrics printed on --once

    final = ObjectStore.load(state)
    pods = final.list("Pod")
    assert len(pods) == 2
    assert all(p.node_name for p in pods)


def test_metrics_export_reference_names():
    METRICS.reset()
    METRICS.observe("e2e_scheduling_latency", 0.01)
    METRICS.observe("action_scheduling_latency:allocate", 0.002)
    METRICS.inc("schedule_attempts_total")
    text = METRICS.export_text()
    assert "e2e_scheduling_latency_p99" in text
    assert "action_scheduling_latency:allocate_mean" in text
    assert "schedule_attempts_total 1.0" in text


def test_prometheus_metrics_source():
    """PrometheusSource against a stub /api/v1/query endpoint."""
    import json as _json
    import threading
    from http.server import BaseHTTPRequestHandler, HTTPServer

    from volcano_amd.scheduler.metrics_source import PrometheusSource

    class Handler(BaseHTTPRequestHandler):
        def do_GET(self):
            val = "77.5" if "cpu" in self.path else "33.0"
            body = _json.dumps({"data": {"result": [
                {"value": [0, val]}]}}).encode()
            self.send_response(200)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def log_message(self, *a):
            pass

    srv = HTTPServer(("127.0.0.1", 0), Handler)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        src = PrometheusSource(f"http://127.0.0.1:{srv.server_address[1]}")
        nodes = [synth.make_node("n1")]
        usage = src.node_usage(nodes)
        assert usage["n1"]["cpu"] == 77.5
        assert usage["n1"]["memory"] == 33.0
    finally:
        srv.shutdown()


def test_cycle_recorder_writes_jsonl(tmp_path, monkeypatch):
    import json
    from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                       default_config)
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth
    log = tmp_path / "cycles.jsonl"
    monkeypatch.setenv("VAMD_CYCLE_LOG", str(log))
    store = ObjectStore()
    cache = SchedulerCache(store=store, binder=FakeBinder())
    sched = Scheduler(cache, default_config())
    store.create("Node", synth.make_node("n0"))
    synth.make_gang(store, "g", replicas=2, cpu_milli=100)
    sched.run_once()
    sched.run_once()
    lines = [json.loads(x) for x in log.read_text().splitlines()]
    assert len(lines) == 2
    assert lines[0]["cycle"] == 1 and lines[1]["cycle"] == 2
    assert "allocate" in lines[0]["actions_ms"]
    assert lines[0]["nodes"] == 1 and lines[0]["jobs"] >= 1
    assert lines[1]["pending_after"] == 0        # gang placed
