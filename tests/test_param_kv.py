"""Versioned parameter-server KV protocol (reference crates/messages
lib.rs:698-739 parameter_pull/parameter_push — declared in the reference's
wire protocol; implemented here as a working store on the worker daemon):
push assigns monotonic versions, pull returns latest or a named version,
missing keys answer found=false (NotFound)."""

import os
import socket
import subprocess
import sys
import time
from pathlib import Path

import pytest

core = pytest.importorskip("hypha_amd._core")
REPO = Path(__file__).resolve().parent.parent
BIN = REPO / "bin"


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.mark.timeout(120)
def test_param_kv_push_pull(tmp_path):
    if not (BIN / "hypha-worker").exists():
        pytest.skip("daemon binaries not built")
    gw_port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    log = open(tmp_path / "daemons.log", "w")
    procs = [subprocess.Popen([str(BIN / "hypha-gateway"), "--port", str(gw_port)],
                              cwd=REPO, env=env, stdout=log, stderr=log)]
    time.sleep(0.3)
    procs.append(subprocess.Popen(
        [str(BIN / "hypha-worker"), "--name", "kv-worker",
         "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
         "--exec-cmd", "true", "--work-root", str(tmp_path / "work")],
        cwd=REPO, env=env, stdout=log, stderr=log))
    client = core.Node("kv-client", "127.0.0.1", gw_port)
    try:
        deadline = time.time() + 10
        while time.time() < deadline:
            try:
                client.start(0)
                break
            except RuntimeError:
                time.sleep(0.2)

        # push v1, v2 (auto-versioned), then an explicit v9
        r, _ = client.stream_call("kv-worker", "param_push",
                                  {"job": "j1", "key": "theta"}, b"AAAA")
        assert r["ok"] and r["version"] == 1
        r, _ = client.stream_call("kv-worker", "param_push",
                                  {"job": "j1", "key": "theta"}, b"BBBB")
        assert r["version"] == 2
        r, _ = client.stream_call("kv-worker", "param_push",
                                  {"job": "j1", "key": "theta", "version": 9}, b"IIII")
        assert r["version"] == 9

        # pull latest
        r, blob = client.stream_call("kv-worker", "param_pull",
                                     {"job": "j1", "key": "theta"})
        assert r["found"] and r["version"] == 9 and blob == b"IIII"
        # pull a named version
        r, blob = client.stream_call("kv-worker", "param_pull",
                                     {"job": "j1", "key": "theta", "version": 1})
        assert r["found"] and blob == b"AAAA"
        # NotFound: unknown key and unknown version
        r, blob = client.stream_call("kv-worker", "param_pull",
                                     {"job": "j1", "key": "nope"})
        assert not r["found"] and blob is None
        r, _ = client.stream_call("kv-worker", "param_pull",
                                  {"job": "j1", "key": "theta", "version": 3})
        assert not r["found"]
        # jobs are namespaced
        r, _ = client.stream_call("kv-worker", "param_pull",
                                  {"job": "j2", "key": "theta"})
        assert not r["found"]
    finally:
        client.stop()
        for p in procs:
            p.terminate()
        for p in procs:
            p.wait(timeout=10)
        log.close()


@pytest.mark.timeout(120)
def test_param_kv_concurrent_stress(tmp_path):
    """Concurrent pushes/pulls from multiple threads: the per-connection
    server threads and the versioned store must not race or drop data."""
    import threading

    gw_port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    log = open(tmp_path / "daemons.log", "w")
    procs = [subprocess.Popen([str(BIN / "hypha-gateway"), "--port", str(gw_port)],
                              cwd=REPO, env=env, stdout=log, stderr=log)]
    time.sleep(0.3)
    procs.append(subprocess.Popen(
        [str(BIN / "hypha-worker"), "--name", "kv-stress",
         "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
         "--exec-cmd", "true", "--work-root", str(tmp_path / "work")],
        cwd=REPO, env=env, stdout=log, stderr=log))
    client = core.Node("kv-stress-client", "127.0.0.1", gw_port)
    try:
        deadline = time.time() + 10
        while time.time() < deadline:
            try:
                client.start(0)
                break
            except RuntimeError:
                time.sleep(0.2)

        errors = []

        def hammer(tid):
            try:
                payload = bytes([tid]) * 4096
                for i in range(20):
                    r, _ = client.stream_call(
                        "kv-stress", "param_push",
                        {"job": "stress", "key": f"k{tid}"}, payload)
                    assert r["ok"] and r["version"] == i + 1, r
                r, blob = client.stream_call(
                    "kv-stress", "param_pull", {"job": "stress", "key": f"k{tid}"})
                assert r["found"] and r["version"] == 20 and blob == payload
            except Exception as e:  # surfaced in the main thread
                errors.append((tid, repr(e)))

        threads = [threading.Thread(target=hammer, args=(t,)) for t in range(8)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=60)
        assert not errors, errors
    finally:
        client.stop()
        for p in procs:
            p.terminate()
        for p in procs:
            p.wait(timeout=10)
        log.close()
