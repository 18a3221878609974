"""Inference-executor test: run the real subprocess against a minimal Python
bridge stub (HTTP over UDS, the same contract the C++ bridge serves)."""

import json
import os
import socketserver
import subprocess
import sys
import threading
from http.server import BaseHTTPRequestHandler
from pathlib import Path

import pytest
import torch

REPO = Path(__file__).resolve().parent.parent


class UDSHttpServer(socketserver.ThreadingUnixStreamServer):
    daemon_threads = True


def test_infer_executor_end_to_end(tmp_path):
    from safetensors.torch import load_file

    from hypha_amd.data.synthetic import write_slice_files

    slice_paths = write_slice_files(str(tmp_path / "slices"), "p", 1, 4, 512, 32)
    statuses = []

    class Handler(BaseHTTPRequestHandler):
        def _json(self, obj):
            body = json.dumps(obj).encode()
            self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def do_POST(self):
            n = int(self.headers.get("Content-Length", 0))
            payload = json.loads(self.rfile.read(n) or b"{}")
            if self.path == "/resources/fetch":
                self._json({"files": slice_paths})
            elif self.path == "/status/send":
                statuses.append(payload)
                self._json({"kind": "ok"})
            else:
                self._json({})

        def log_message(self, *a):
            pass

    sock_path = str(tmp_path / "bridge.sock")
    server = UDSHttpServer(sock_path, Handler)
    threading.Thread(target=server.serve_forever, daemon=True).start()

    work = tmp_path / "work"
    work.mkdir()
    job = tmp_path / "job.json"
    job.write_text(json.dumps({
        "model": "llama-tiny", "data": {"any": "ref"},
        "batch_size": 2, "seq_len": 16, "max_new_tokens": 4, "num_batches": 2,
    }))
    r = subprocess.run(
        [sys.executable, "-m", "hypha_amd.runtime.infer_executor",
         "--socket", sock_path, "--work-dir", str(work), "--job", str(job)],
        cwd=REPO, env=dict(os.environ, PYTHONPATH=str(REPO)),
        capture_output=True, text=True, timeout=180,
    )
    server.shutdown()
    assert r.returncode == 0, r.stderr[-2000:]
    outs = sorted(work.glob("completion-*.safetensors"))
    assert len(outs) == 2
    tokens = load_file(str(outs[0]))["tokens"]
    assert tokens.shape == (2, 20)  # 16 prompt + 4 generated
    assert any(s.get("kind") == "metrics" for s in statuses)
