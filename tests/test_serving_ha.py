"""Serving HA: kill/restart failover across two replica processes.

Torch-free reference scenario: c_api_ha_test.cpp:156-212 SIGKILLs servers
while readers keep succeeding through replica failover, then restarts them.
Here: two serving processes load the same dump; the ServingClient pulls
through both; one replica is SIGKILLed mid-stream (reads must keep
succeeding via the survivor); the dead replica is restarted and must serve
again."""

import os
import signal
import socket
import subprocess
import sys
import time

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _start_server(port, dump):
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    return subprocess.Popen(
        [sys.executable, "-m", "openembedding_amd.serving",
         "--host", "127.0.0.1", "--port", str(port),
         "--device", "cpu", "--model-uri", dump],
        cwd=REPO, env=env,
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)


def _start_empty_server(port):
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    return subprocess.Popen(
        [sys.executable, "-m", "openembedding_amd.serving",
         "--host", "127.0.0.1", "--port", str(port), "--device", "cpu"],
        cwd=REPO, env=env,
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)


def _wait_http(port, timeout=60):
    import requests
    t0 = time.time()
    while time.time() - t0 < timeout:
        try:
            r = requests.get(f"http://127.0.0.1:{port}/models", timeout=2)
            if r.status_code == 200:
                return True
        except Exception:  # noqa: BLE001
            pass
        time.sleep(0.3)
    return False


def _wait_ready(port, timeout=60):
    import requests
    t0 = time.time()
    while time.time() - t0 < timeout:
        try:
            r = requests.get(f"http://127.0.0.1:{port}/models", timeout=2)
            if r.status_code == 200 and r.json():
                return True
        except Exception:  # noqa: BLE001
            pass
        time.sleep(0.3)
    return False


@pytest.mark.timeout(300)
def test_failover_kill_restart(tmp_path):
    from openembedding_amd import checkpoint
    from openembedding_amd.context import Context
    from openembedding_amd.serving import ServingClient

    # build a small trained dump
    ctx = Context(device="cpu")
    st = ctx.create_storage()
    var = st.create_variable(500, 6)
    var.set_initializer("uniform", minval=-1, maxval=1)
    var.set_optimizer("adagrad", learning_rate=0.1)
    keys = torch.arange(0, 200, dtype=torch.int64)
    out, h = var.pull(keys)
    var.push(h, torch.ones_like(out))
    st.update_weights()
    expected, _ = var.pull(keys, readonly=True)
    dump = str(tmp_path / "dump")
    checkpoint.dump_model(ctx, dump)
    sign = checkpoint.read_meta(dump)["model_sign"]
    vid = var.variable_id

    ports = [_free_port(), _free_port()]
    procs = [_start_server(p, dump) for p in ports]
    try:
        for p in ports:
            assert _wait_ready(p), f"server on {p} did not become ready"
        client = ServingClient([f"http://127.0.0.1:{p}" for p in ports],
                               timeout=5.0)

        probe = keys[:32]
        want = expected[:32]

        def check_pull():
            w = torch.tensor(client.pull(sign, vid, probe))
            torch.testing.assert_close(w, want, rtol=1e-5, atol=1e-6)

        for _ in range(4):
            check_pull()          # both replicas serving

        # SIGKILL replica 0 (reference killer thread); reads must keep
        # succeeding through the survivor
        procs[0].send_signal(signal.SIGKILL)
        procs[0].wait(timeout=30)
        for _ in range(6):
            check_pull()

        # restart-and-reload the dead replica; it must serve again
        procs[0] = _start_server(ports[0], dump)
        assert _wait_ready(ports[0])
        solo = ServingClient([f"http://127.0.0.1:{ports[0]}"])
        w = torch.tensor(solo.pull(sign, vid, probe))
        torch.testing.assert_close(w, want, rtol=1e-5, atol=1e-6)
        for _ in range(4):
            check_pull()

        # coordinated restore: a THIRD replica reconstructs the model from
        # a LIVE replica over the wire — no dump involved (reference
        # EmbeddingRestoreOperator.cpp:19-106 replica path)
        import requests
        p3 = _free_port()
        proc3 = _start_empty_server(p3)
        procs.append(proc3)
        assert _wait_http(p3)
        r = requests.post(
            f"http://127.0.0.1:{p3}/models",
            json={"from_replica": f"http://127.0.0.1:{ports[1]}",
                  "sign": sign}, timeout=120)
        assert r.status_code == 200, r.text
        restored = ServingClient([f"http://127.0.0.1:{p3}"])
        w3 = torch.tensor(restored.pull(sign, vid, probe))
        torch.testing.assert_close(w3, want, rtol=1e-5, atol=1e-6)
    finally:
        for p in procs:
            if p.poll() is None:
                p.kill()
                p.wait(timeout=30)
