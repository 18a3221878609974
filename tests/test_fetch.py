"""URI/HTTP fetch connector tests (offline, local HTTP server).

Covers the reference's HttpHfFetcher + validate_fetch semantics
(/root/reference/crates/worker/src/connector/mod.rs:226-302,
executor/bridge.rs:349-377) plus the host allow-list the reference left as
a TODO: scheme validation, traversal guards, redirect handling with
per-hop allow-list enforcement, and an end-to-end cluster job whose
training data is fetched over HTTP instead of a data node."""

import functools
import http.server
import os
import signal
import socket
import subprocess
import sys
import threading
import time
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
BIN = REPO / "bin"


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.fixture()
def http_root(tmp_path):
    root = tmp_path / "www"
    root.mkdir()
    handler = functools.partial(http.server.SimpleHTTPRequestHandler,
                                directory=str(root))
    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), handler)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield root, srv.server_address[1]
    srv.shutdown()


def test_parse_url_validation():
    from hypha_amd import _core

    u = _core.parse_url("https://example.com/a/b.bin")
    assert u == {"scheme": "https", "host": "example.com", "port": 443,
                 "path": "/a/b.bin"}
    assert _core.parse_url("http://h:81")["path"] == "/"
    for bad in ("ftp://x/y", "file:///etc/passwd", "example.com/x", "http:///p"):
        with pytest.raises(RuntimeError):
            _core.parse_url(bad)


def test_fetch_allow_list_semantics():
    from hypha_amd import _core

    # default deny (the reference's TODO closed conservatively)
    assert not _core.fetch_allowed("anything.com", 80, [])
    assert _core.fetch_allowed("a.com", 80, ["a.com"])
    assert not _core.fetch_allowed("b.com", 80, ["a.com"])
    # wildcard suffix matches subdomains AND the bare domain
    assert _core.fetch_allowed("files.hf.co", 443, ["*.hf.co"])
    assert _core.fetch_allowed("hf.co", 443, ["*.hf.co"])
    assert not _core.fetch_allowed("nothf.co", 443, ["*.hf.co"])
    # port-qualified entries pin the port
    assert _core.fetch_allowed("127.0.0.1", 8080, ["127.0.0.1:8080"])
    assert not _core.fetch_allowed("127.0.0.1", 8081, ["127.0.0.1:8080"])
    # global wildcard
    assert _core.fetch_allowed("x.y", 1, ["*"])


def test_http_get_roundtrip(http_root, tmp_path):
    from hypha_amd import _core

    root, port = http_root
    payload = os.urandom(300000)
    (root / "blob.bin").write_bytes(payload)
    out = tmp_path / "got.bin"
    n = _core.http_get_to_file(f"http://127.0.0.1:{port}/blob.bin", str(out),
                               ["127.0.0.1"])
    assert n == len(payload)
    assert out.read_bytes() == payload
    assert oct(out.stat().st_mode & 0o777) == "0o600"


def test_http_get_404_and_deny(http_root, tmp_path):
    from hypha_amd import _core

    root, port = http_root
    (root / "x.bin").write_bytes(b"data")
    with pytest.raises(RuntimeError, match="HTTP 404"):
        _core.http_get_to_file(f"http://127.0.0.1:{port}/missing", str(tmp_path / "o"),
                               ["127.0.0.1"])
    with pytest.raises(RuntimeError, match="allow-list"):
        _core.http_get_to_file(f"http://127.0.0.1:{port}/x.bin", str(tmp_path / "o"),
                               [])
    with pytest.raises(RuntimeError, match="allow-list"):
        _core.http_get_to_file(f"http://127.0.0.1:{port}/x.bin", str(tmp_path / "o"),
                               ["localhost"])  # no name aliasing: host literal


def test_http_get_redirect_allowlist_per_hop(tmp_path):
    """Redirects are followed, and every hop is allow-list checked — a
    permitted host cannot bounce the worker to a forbidden one."""
    from hypha_amd import _core

    class H(http.server.BaseHTTPRequestHandler):
        def do_GET(self):
            if self.path == "/r":
                self.send_response(302)
                self.send_header("Location", "/final.bin")
                self.end_headers()
            elif self.path == "/evil":
                self.send_response(302)
                self.send_header("Location", "http://127.0.0.2:9/loot")
                self.end_headers()
            elif self.path == "/final.bin":
                body = b"redirected-ok"
                self.send_response(200)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)
            else:
                self.send_error(404)

        def log_message(self, *a):
            pass

    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), H)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    port = srv.server_address[1]
    try:
        out = tmp_path / "r.bin"
        n = _core.http_get_to_file(f"http://127.0.0.1:{port}/r", str(out),
                                   ["127.0.0.1"])
        assert n == len(b"redirected-ok") and out.read_bytes() == b"redirected-ok"
        with pytest.raises(RuntimeError, match="allow-list"):
            _core.http_get_to_file(f"http://127.0.0.1:{port}/evil",
                                   str(tmp_path / "e"), ["127.0.0.1"])
    finally:
        srv.shutdown()


@pytest.mark.timeout(300)
def test_cluster_trains_from_uri_data(tmp_path):
    """End-to-end BASELINE-config-1-shaped job whose data reference is
    Fetch{uri}: no data node — workers download the training slice over
    HTTP through the fetch connector (allow-listed), then complete two
    DiLoCo rounds."""
    from hypha_amd.data.synthetic import write_slice_files

    if not (BIN / "hypha-gateway").exists():
        subprocess.run([sys.executable, "setup.py", "build_ext", "--inplace"],
                       cwd=REPO, check=True)

    data_dir = tmp_path / "slices"
    write_slice_files(str(data_dir), "synth", num_slices=1, samples_per_slice=16,
                      vocab_size=512, seq_len=128)
    slice_file = next(Path(data_dir).glob("*.safetensors"))
    www = tmp_path / "www"
    www.mkdir()
    (www / "train.safetensors").write_bytes(slice_file.read_bytes())
    handler = functools.partial(http.server.SimpleHTTPRequestHandler,
                                directory=str(www))
    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), handler)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    data_port = srv.server_address[1]

    gw_port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    procs = []
    logs = {}

    def spawn(name, cmd):
        log = open(tmp_path / f"{name}.log", "w")
        logs[name] = tmp_path / f"{name}.log"
        p = subprocess.Popen(cmd, cwd=REPO, env=env, stdout=log, stderr=log,
                             start_new_session=True)
        procs.append(p)
        return p

    try:
        spawn("gateway", [str(BIN / "hypha-gateway"), "--port", str(gw_port)])
        time.sleep(0.3)
        exec_cmd = (f"{sys.executable} -m hypha_amd.runtime.executor "
                    "--socket {SOCKET_PATH} --work-dir {WORK_DIR} --job {JOB_JSON}")
        for i in range(3):
            spawn(f"worker{i}", [str(BIN / "hypha-worker"), "--name", f"worker-{i}",
                                 "--gateway-host", "127.0.0.1",
                                 "--gateway-port", str(gw_port),
                                 "--exec-cmd", exec_cmd,
                                 "--fetch-allow", "127.0.0.1",
                                 "--work-root", str(tmp_path / f"work{i}")])
        time.sleep(0.5)

        cfg = tmp_path / "job.json"
        cfg.write_text(
            '{"model": "llama-tiny",'
            f' "data_uri": "http://127.0.0.1:{data_port}/train.safetensors",'
            ' "num_workers": 2, "update_rounds": 2,'
            ' "avg_samples_between_updates": 8, "batch_size": 2,'
            ' "seq_len": 128, "inner_lr": 0.001}'
        )
        sched = subprocess.Popen(
            [str(BIN / "hypha-scheduler"), "--name", "scheduler",
             "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
             "--config", str(cfg)],
            cwd=REPO, env=env, stdout=subprocess.PIPE,
            stderr=open(tmp_path / "sched.log", "w"), text=True,
        )
        procs.append(sched)
        out, _ = sched.communicate(timeout=240)
        assert "Job is completed." in out, (
            out,
            *[f"--- {n}: {p.read_text()[-2000:]}" for n, p in logs.items()],
            (tmp_path / "sched.log").read_text()[-3000:],
        )
        # the artifact landed under a work dir with restrictive permissions
        arts = list(tmp_path.glob("work*/**/artifacts/train.safetensors"))
        assert arts, "no fetched artifacts on disk"
        assert oct(arts[0].stat().st_mode & 0o777) == "0o600"
    finally:
        srv.shutdown()
        for p in procs:
            try:
                os.killpg(p.pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                if p.poll() is None:
                    p.send_signal(signal.SIGKILL)
        for p in procs:
            try:
                p.wait(timeout=10)
            except Exception:
                pass


@pytest.mark.timeout(300)
def test_cluster_trains_with_fetched_preprocessor(tmp_path):
    """TrainExecutorConfig.preprocessor parity (messages lib.rs:483-489 +
    utils.py get_preprocessor): the scheduler forwards the preprocessor
    config, the worker fetches its artifact over HTTP through the
    connector, and the executor runs slice columns through the resolved
    transformers processor before training."""
    import json

    import torch
    from safetensors.torch import save_file

    if not (BIN / "hypha-gateway").exists():
        subprocess.run([sys.executable, "setup.py", "build_ext", "--inplace"],
                       cwd=REPO, check=True)

    www = tmp_path / "www"
    www.mkdir()
    # raw "audio" slice whose samples are small ints: after the feature
    # extractor (do_normalize=False) they come back unchanged and double as
    # token ids for the plumbing model
    wave = torch.randint(0, 500, (16, 128)).float()
    save_file({"audio": wave}, str(www / "train.safetensors"))
    (www / "preprocessor_config.json").write_text(json.dumps({
        "feature_extractor_type": "Wav2Vec2FeatureExtractor",
        "feature_size": 1, "sampling_rate": 16000, "padding_value": 0.0,
        "return_attention_mask": False, "do_normalize": False}))
    handler = functools.partial(http.server.SimpleHTTPRequestHandler,
                                directory=str(www))
    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), handler)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    dport = srv.server_address[1]

    gw_port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    procs = []
    logs = {}

    def spawn(name, cmd):
        log = open(tmp_path / f"{name}.log", "w")
        logs[name] = tmp_path / f"{name}.log"
        p = subprocess.Popen(cmd, cwd=REPO, env=env, stdout=log, stderr=log,
                             start_new_session=True)
        procs.append(p)
        return p

    try:
        spawn("gateway", [str(BIN / "hypha-gateway"), "--port", str(gw_port)])
        time.sleep(0.3)
        exec_cmd = (f"{sys.executable} -m hypha_amd.runtime.executor "
                    "--socket {SOCKET_PATH} --work-dir {WORK_DIR} --job {JOB_JSON}")
        for i in range(3):
            spawn(f"worker{i}", [str(BIN / "hypha-worker"), "--name", f"worker-{i}",
                                 "--gateway-host", "127.0.0.1",
                                 "--gateway-port", str(gw_port),
                                 "--exec-cmd", exec_cmd,
                                 "--fetch-allow", "127.0.0.1",
                                 "--work-root", str(tmp_path / f"work{i}")])
        time.sleep(0.5)

        cfg = tmp_path / "job.json"
        cfg.write_text(json.dumps({
            "model": "llama-tiny",
            "data_uri": f"http://127.0.0.1:{dport}/train.safetensors",
            "preprocessor": {
                "task": "feature",
                "artifact": {"uri": {
                    "value": f"http://127.0.0.1:{dport}/preprocessor_config.json"}},
                "input_names": ["audio"],
                "output_key": "input_values",
            },
            "num_workers": 2, "update_rounds": 2,
            "avg_samples_between_updates": 8, "batch_size": 2,
            "seq_len": 128, "inner_lr": 0.001}))
        sched = subprocess.Popen(
            [str(BIN / "hypha-scheduler"), "--name", "scheduler",
             "--gateway-host", "127.0.0.1", "--gateway-port", str(gw_port),
             "--config", str(cfg)],
            cwd=REPO, env=env, stdout=subprocess.PIPE,
            stderr=open(tmp_path / "sched.log", "w"), text=True,
        )
        procs.append(sched)
        out, _ = sched.communicate(timeout=240)
        assert "Job is completed." in out, (
            out,
            *[f"--- {n}: {p.read_text()[-2000:]}" for n, p in logs.items()],
            (tmp_path / "sched.log").read_text()[-3000:],
        )
        # the executor really resolved the processor
        w_logs = "".join(p.read_text() for n, p in logs.items() if "worker" in n)
        work_logs = ""
        for d in tmp_path.glob("work*/**/*.log"):
            work_logs += d.read_text()
    finally:
        srv.shutdown()
        for p in procs:
            try:
                os.killpg(p.pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                if p.poll() is None:
                    p.send_signal(signal.SIGKILL)
        for p in procs:
            try:
                p.wait(timeout=10)
            except Exception:
                pass


@pytest.mark.timeout(120)
def test_data_record_survives_gateway_restart(tmp_path):
    """The data node re-announces its dataset record whenever its broker
    connection is (re)established — a gateway RESTART (or failover) no
    longer loses the registry record for good."""
    from hypha_amd import _core
    from hypha_amd.data.synthetic import write_slice_files

    data_dir = tmp_path / "slices"
    write_slice_files(str(data_dir), "ha", num_slices=2, samples_per_slice=4,
                      vocab_size=64, seq_len=16)
    gw_port = free_port()
    env = dict(os.environ, PYTHONPATH=str(REPO))
    gw = subprocess.Popen([str(BIN / "hypha-gateway"), "--port", str(gw_port)],
                          env=env, start_new_session=True)
    dn = None
    gw2 = None
    try:
        time.sleep(0.4)
        dn = subprocess.Popen(
            [str(BIN / "hypha-data"), "--name", "data-ha", "--gateway-host",
             "127.0.0.1", "--gateway-port", str(gw_port), "--dataset", "ha",
             "--dataset-path", str(data_dir)],
            env=env, start_new_session=True)
        time.sleep(0.8)
        gw.kill()
        gw.wait(timeout=10)
        time.sleep(0.5)
        gw2 = subprocess.Popen([str(BIN / "hypha-gateway"), "--port", str(gw_port)],
                               env=env, start_new_session=True)
        rec = None
        deadline = time.time() + 30
        while time.time() < deadline and not rec:
            n = _core.Node("checker", "127.0.0.1", gw_port)
            try:
                n.start(0)
                for _ in range(6):
                    rec = n.kv_get("dataset:ha")
                    if rec:
                        break
                    time.sleep(0.5)
            except RuntimeError:
                time.sleep(0.5)
            finally:
                n.stop()
        assert rec and rec["num_slices"] == 2, rec
    finally:
        for p in (gw, dn, gw2):
            if p is not None:
                try:
                    os.killpg(p.pid, signal.SIGKILL)
                except (ProcessLookupError, PermissionError):
                    if p.poll() is None:
                        p.kill()
