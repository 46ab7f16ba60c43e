"""Native daemon integration test: coordd + keystoned + workerd + bbctl +
bb_bench as real processes over TCP loopback (the start_cluster.sh path)."""
import json
import os
import signal
import socket
import subprocess
import time
import urllib.request

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BIN = os.path.join(REPO, "bin")


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.fixture(scope="module")
def daemon_cluster(tmp_path_factory):
    if not os.path.exists(os.path.join(BIN, "coordd")):
        pytest.skip("daemons not built")
    tmp = tmp_path_factory.mktemp("cluster")
    coord_port = free_port()
    ks_port = free_port()
    metrics_port = free_port()
    procs = []

    def spawn(args, log):
        f = open(tmp / log, "w")
        p = subprocess.Popen(args, stdout=f, stderr=subprocess.STDOUT)
        procs.append(p)
        return p

    spawn([f"{BIN}/coordd", "--listen-host", "127.0.0.1",
           "--listen-port", str(coord_port)], "coordd.log")
    time.sleep(0.3)
    spawn([f"{BIN}/keystoned",
           "--listen-address", f"127.0.0.1:{ks_port}",
           "--coord-endpoint", f"127.0.0.1:{coord_port}",
           "--metrics-address", f"127.0.0.1:{metrics_port}"], "keystoned.log")
    time.sleep(0.3)
    cfg = {
        "worker_id": "dw0",
        "coord_endpoint": f"127.0.0.1:{coord_port}",
        "data_listen_address": "127.0.0.1:0",
        "pools": [{"pool_id": "dpool0", "storage_class": "RAM_CPU",
                   "size_bytes": 64 << 20}],
    }
    cfg_path = tmp / "worker.json"
    cfg_path.write_text(json.dumps(cfg))
    spawn([f"{BIN}/workerd", "--config", str(cfg_path)], "workerd.log")

    # wait for assembly
    deadline = time.time() + 10
    while time.time() < deadline:
        r = subprocess.run([f"{BIN}/bbctl", "--keystone",
                            f"127.0.0.1:{ks_port}", "stat"],
                           capture_output=True, text=True)
        if r.returncode == 0 and "workers=1" in r.stdout:
            break
        time.sleep(0.2)
    else:
        for p in procs:
            p.send_signal(signal.SIGTERM)
        logs = {f: (tmp / f).read_text()[-500:] for f in
                ["coordd.log", "keystoned.log", "workerd.log"]}
        pytest.fail(f"cluster did not assemble: {logs}")

    yield {"keystone": f"127.0.0.1:{ks_port}", "metrics": metrics_port,
           "tmp": tmp}
    for p in procs:
        p.send_signal(signal.SIGTERM)
    for p in procs:
        try:
            p.wait(timeout=5)
        except subprocess.TimeoutExpired:
            p.kill()


def bbctl(cluster, *args, data=None):
    return subprocess.run(
        [f"{BIN}/bbctl", "--keystone", cluster["keystone"], *args],
        capture_output=True, input=data)


class TestDaemons:
    def test_put_get_roundtrip(self, daemon_cluster):
        payload = os.urandom(100 * 1024)
        r = bbctl(daemon_cluster, "put", "cli-obj", "-", data=payload)
        assert r.returncode == 0, r.stderr
        r = bbctl(daemon_cluster, "get", "cli-obj")
        assert r.returncode == 0 and r.stdout == payload
        r = bbctl(daemon_cluster, "verify", "cli-obj")
        assert r.returncode == 0, r.stderr
        r = bbctl(daemon_cluster, "rm", "cli-obj")
        assert r.returncode == 0

    def test_workers_pools_listing(self, daemon_cluster):
        r = bbctl(daemon_cluster, "workers")
        assert r.returncode == 0 and b"dw0" in r.stdout
        r = bbctl(daemon_cluster, "pools")
        assert r.returncode == 0 and b"dpool0" in r.stdout
        assert b"RAM_CPU" in r.stdout

    def test_metrics_endpoint(self, daemon_cluster):
        url = f"http://127.0.0.1:{daemon_cluster['metrics']}/metrics"
        body = urllib.request.urlopen(url, timeout=5).read().decode()
        assert "blackbird_workers 1" in body
        assert "blackbird_pool_capacity_bytes" in body
        assert "blackbird_repairs_total" in body
        assert "blackbird_scrub_quarantined_total" in body
        assert "blackbird_token_commits_total" in body
        assert "blackbird_is_leader 1" in body
        stats = json.loads(urllib.request.urlopen(
            f"http://127.0.0.1:{daemon_cluster['metrics']}/stats",
            timeout=5).read())
        assert stats["num_workers"] == 1
        health = urllib.request.urlopen(
            f"http://127.0.0.1:{daemon_cluster['metrics']}/healthz",
            timeout=5).read()
        assert health == b"ok\n"

    def test_bb_bench_runs(self, daemon_cluster):
        r = subprocess.run(
            [f"{BIN}/bb_bench", "--keystone", daemon_cluster["keystone"],
             "--size", "65536", "--iters", "4", "--batch", "8"],
            capture_output=True, text=True, timeout=120)
        assert r.returncode == 0, r.stderr
        assert "WRITE:" in r.stdout and "READ:" in r.stdout


class TestAdminCommands:
    def test_bbctl_maintenance_triggers(self, daemon_cluster):
        """Admin RPCs through the CLI: scrub / repair / compact run against
        the live daemons and report their results."""
        ks = daemon_cluster["keystone"]
        r = subprocess.run([f"{BIN}/bbctl", "--keystone", ks, "put",
                            "admin-obj", "-"], input=b"x" * 65536,
                           capture_output=True)
        assert r.returncode == 0, r.stderr
        r = subprocess.run([f"{BIN}/bbctl", "--keystone", ks, "scrub"],
                           capture_output=True, text=True, timeout=30)
        assert r.returncode == 0 and "quarantined 0" in r.stdout, r.stdout
        r = subprocess.run([f"{BIN}/bbctl", "--keystone", ks, "repair"],
                           capture_output=True, text=True, timeout=30)
        assert r.returncode == 0 and "repair pass done" in r.stdout
        pools = subprocess.run([f"{BIN}/bbctl", "--keystone", ks, "pools"],
                               capture_output=True, text=True).stdout
        pool_id = pools.split("\t")[0]
        r = subprocess.run([f"{BIN}/bbctl", "--keystone", ks, "compact",
                            pool_id], capture_output=True, text=True,
                           timeout=30)
        assert r.returncode == 0 and "moved" in r.stdout, r.stdout
