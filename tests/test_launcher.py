"""Full multi-process integration: the launcher hosts apiserver +
controllers + scheduler; vcctl drives it over HTTP from this process."""

import socket
import subprocess
import sys
import time

import pytest

from volcano_amd.api.objects import to_dict
from volcano_amd.store.client import StoreClient
from volcano_amd.utils import synth

GI = 1024 ** 3


def test_launcher_end_to_end(tmp_path):
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    state = str(tmp_path / "cluster.json")
    proc = subprocess.Popen(
        [sys.executable, "-m", "volcano_amd.launcher", "--state", state,
         "--api-port", str(port), "--period", "0.1"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    client = StoreClient(f"http://127.0.0.1:{port}")
    try:
        for _ in range(200):
            if client.healthz():
                break
            time.sleep(0.1)
        else:
            out = proc.communicate(timeout=5)[0]
            pytest.fail(f"launcher did not come up: {out!r}")

        for n in synth.make_nodes(3, cpu_milli=8000, mem=32 * GI):
            client.create("Node", n)

        from volcano_amd.cli.vcctl import main as vcctl
        url = f"http://127.0.0.1:{port}"
        assert vcctl(["--server", url, "queue", "create", "-N", "prod",
                      "-w", "2"]) == 0
        assert vcctl(["--server", url, "job", "run", "-N", "train",
                      "-r", "4", "-q", "prod", "--cpu", "1"]) == 0

        # controller expands, scheduler binds — all in the launcher process
        deadline = time.time() + 20
        while time.time() < deadline:
            pods = client.list("Pod")
            if len(pods) == 4 and all(p.node_name for p in pods):
                break
            time.sleep(0.2)
        else:
            pytest.fail(f"pods not scheduled: "
                        f"{[(p.meta.name, p.node_name) for p in client.list('Pod')]}")

        pg = client.get("PodGroup", "default", "train")
        assert pg is not None and pg.status.phase in ("Inqueue", "Running")
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
