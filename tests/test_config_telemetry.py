"""Config layering/validation + telemetry registry tests (SURVEY.md §5:
config crate and telemetry crate parity)."""

import json

import pytest

from hypha_amd.config import ConfigError, JobConfig, example_config, load_job_config
from hypha_amd.telemetry import Metrics


def test_defaults_validate():
    cfg = load_job_config()
    assert cfg.model == "llama3-8b" and cfg.num_workers == 8


def test_file_layer(tmp_path):
    p = tmp_path / "job.json"
    p.write_text(json.dumps({"model": "gpt2-small", "num_workers": 2}))
    cfg = load_job_config(str(p))
    assert cfg.model == "gpt2-small" and cfg.num_workers == 2
    assert cfg.seq_len == 2048  # default survives


def test_toml_layer(tmp_path):
    p = tmp_path / "job.toml"
    p.write_text('model = "llama3-70b"\nbatch_size = 2\n')
    cfg = load_job_config(str(p))
    assert cfg.model == "llama3-70b" and cfg.batch_size == 2


def test_env_overrides_file(tmp_path):
    p = tmp_path / "job.json"
    p.write_text(json.dumps({"num_workers": 2}))
    cfg = load_job_config(str(p), env={"HYPHA_NUM_WORKERS": "4"})
    assert cfg.num_workers == 4


def test_explicit_overrides_env():
    cfg = load_job_config(env={"HYPHA_SEQ_LEN": "512"}, seq_len=1024)
    assert cfg.seq_len == 1024


def test_unknown_key_rejected(tmp_path):
    p = tmp_path / "job.json"
    p.write_text(json.dumps({"no_such_option": 1}))
    with pytest.raises(ConfigError, match="no_such_option"):
        load_job_config(str(p))


def test_validation_errors():
    with pytest.raises(ConfigError, match="num_workers"):
        JobConfig(num_workers=0).validate()
    with pytest.raises(ConfigError, match="lr_schedule"):
        JobConfig(lr_schedule="exotic").validate()


def test_example_config_mentions_every_field():
    text = example_config()
    for f in ("model", "outer_lr", "update_rounds", "seq_len"):
        assert f'"{f}"' in text


def test_metrics_counters_and_gauges():
    m = Metrics()
    m.counter_add("a.b", 2)
    m.counter_add("a.b", 3)
    m.gauge_set("g", 7.5, rank=0)
    snap = m.snapshot()
    assert snap["counters"]["a.b"] == 5
    assert snap["gauges"]["g{rank=0}"] == 7.5


def test_comm_instrumentation():
    import torch

    from hypha_amd.parallel import Comm
    from hypha_amd.telemetry import METRICS, instrument_comm

    c = Comm()
    instrument_comm(c)
    before = METRICS.snapshot()["counters"].get(
        "hypha.bandwidth.collective.payload_bytes", 0
    )
    c.all_reduce_mean_flat(torch.zeros(1000))
    after = METRICS.snapshot()["counters"]["hypha.bandwidth.collective.payload_bytes"]
    assert after - before == 4000


def test_certutil_pki_roundtrip(tmp_path):
    """3-tier PKI generation + chain verification (certutil crate parity)."""
    import subprocess
    import sys
    from pathlib import Path

    tool = Path(__file__).resolve().parent.parent / "tools" / "hypha_certutil.py"
    out = tmp_path / "pki"
    subprocess.run([sys.executable, str(tool), "root", "--out", str(out)], check=True)
    subprocess.run([sys.executable, str(tool), "org", "--out", str(out),
                    "--name", "org1"], check=True)
    r = subprocess.run([sys.executable, str(tool), "node", "--out", str(out),
                        "--org", "org1", "--name", "worker-0"],
                       check=True, capture_output=True, text=True)
    assert "peer-id: peer-" in r.stdout
    # openssl verifies the chain root -> org -> node
    v = subprocess.run(
        ["openssl", "verify", "-CAfile", str(out / "root.crt"),
         "-untrusted", str(out / "org1.crt"), str(out / "worker-0.crt")],
        capture_output=True, text=True)
    assert v.returncode == 0, v.stderr


def test_aim_driver_endpoint(tmp_path, monkeypatch):
    """drivers/aim_driver.py must accept the scheduler's metric posts
    (metrics_bridge.rs -> aim-driver/main.py parity)."""
    import importlib
    import json as pyjson

    monkeypatch.setenv("HYPHA_AIM_LOG", str(tmp_path / "run.jsonl"))
    import drivers.aim_driver as ad

    importlib.reload(ad)
    from fastapi.testclient import TestClient

    client = TestClient(ad.app)
    r = client.post("/status", json={"worker_id": "w0", "round": 3,
                                     "metric_name": "loss", "value": 1.25})
    assert r.status_code == 200 and r.json() == {"ok": True}
    rec = pyjson.loads((tmp_path / "run.jsonl").read_text().splitlines()[0])
    assert rec["worker_id"] == "w0" and rec["value"] == 1.25


def test_daemon_init_and_probe_subcommands():
    """Reference CLI parity (scheduler/src/bin/hypha-scheduler.rs:459-548
    Init/Probe/Run): every daemon emits a commented config on `init` and
    fails `probe` cleanly when no gateway is listening."""
    import subprocess
    from pathlib import Path

    bins = Path(__file__).resolve().parent.parent / "bin"
    if not (bins / "hypha-gateway").exists():
        pytest.skip("daemon binaries not built")
    for d in ("hypha-gateway", "hypha-worker", "hypha-scheduler", "hypha-data"):
        r = subprocess.run([str(bins / d), "init"], capture_output=True, text=True)
        assert r.returncode == 0 and r.stdout.startswith("#"), d
    for d, args in (("hypha-gateway", ["--port", "59997"]),
                    ("hypha-worker", ["--gateway-port", "59997"]),
                    ("hypha-scheduler", ["--gateway-port", "59997"]),
                    ("hypha-data", ["--gateway-port", "59997"])):
        r = subprocess.run([str(bins / d), "probe", *args],
                           capture_output=True, text=True, timeout=30)
        assert r.returncode == 1 and "unreachable" in r.stderr, d


def test_collect_transport_bytes():
    from hypha_amd.telemetry import METRICS, collect_transport_bytes

    s = collect_transport_bytes()
    if s is None:
        pytest.skip("native core not built")
    assert s["inbound_bytes"] >= 0 and s["outbound_bytes"] >= 0
    snap = METRICS.snapshot()
    assert "hypha.bandwidth.transport.inbound_bytes" in snap["gauges"]


def test_tracer_span_nesting_and_otlp_shape(tmp_path):
    """OTLP-shaped tracing (reference telemetry/src/lib.rs:15-49 parity):
    trace/span id propagation, nesting via the context stack, and the
    OTLP/JSON ResourceSpans schema in the file sink."""
    import json

    from hypha_amd.telemetry import Tracer

    sink = tmp_path / "traces.jsonl"
    tr = Tracer(service_name="test-svc", sink_path=str(sink), flush_every=999)
    with tr.start_span("job.execute", job_id="j1") as job:
        with tr.start_span("diloco.round", round=0) as rnd:
            assert rnd.trace_id == job.trace_id
            assert rnd.parent_id == job.span_id
        with tr.start_span("diloco.outer_sync", round=0) as sync:
            sync.set_attribute("payload_bytes", 123)
    n = tr.flush()
    assert n == 3
    batch = json.loads(sink.read_text().strip())
    rs = batch["resourceSpans"][0]
    svc = rs["resource"]["attributes"][0]
    assert svc["key"] == "service.name"
    assert svc["value"]["stringValue"] == "test-svc"
    spans = rs["scopeSpans"][0]["spans"]
    assert {s["name"] for s in spans} == {"job.execute", "diloco.round",
                                          "diloco.outer_sync"}
    for s in spans:
        assert len(s["traceId"]) == 32 and len(s["spanId"]) == 16
        assert int(s["endTimeUnixNano"]) >= int(s["startTimeUnixNano"])
        assert s["status"]["code"] == 1
    sync_s = next(s for s in spans if s["name"] == "diloco.outer_sync")
    assert {"key": "payload_bytes", "value": {"intValue": "123"}} in sync_s["attributes"]


def test_tracer_posts_otlp_http(tmp_path):
    """Spans POST to an OTLP/HTTP collector endpoint (/v1/traces)."""
    import http.server
    import json
    import threading

    got = {}

    class H(http.server.BaseHTTPRequestHandler):
        def do_POST(self):
            n = int(self.headers["Content-Length"])
            got["path"] = self.path
            got["body"] = json.loads(self.rfile.read(n))
            self.send_response(200)
            self.end_headers()
            self.wfile.write(b"{}")

        def log_message(self, *a):
            pass

    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), H)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    try:
        from hypha_amd.telemetry import Tracer

        tr = Tracer(endpoint=f"http://127.0.0.1:{srv.server_address[1]}")
        with tr.start_span("dispatch", job_id="j2"):
            pass
        assert tr.flush() == 1
        assert tr.export_errors == 0
        assert got["path"] == "/v1/traces"
        names = [s["name"] for s in
                 got["body"]["resourceSpans"][0]["scopeSpans"][0]["spans"]]
        assert names == ["dispatch"]
    finally:
        srv.shutdown()


def test_tracer_marks_exception_spans(tmp_path):
    from hypha_amd.telemetry import Tracer

    tr = Tracer(sink_path=str(tmp_path / "t.jsonl"), flush_every=999)
    try:
        with tr.start_span("failing"):
            raise ValueError("boom")
    except ValueError:
        pass
    import json

    tr.flush()
    span = json.loads((tmp_path / "t.jsonl").read_text())[
        "resourceSpans"][0]["scopeSpans"][0]["spans"][0]
    assert span["status"]["code"] == 2
