"""Mutual-TLS control plane: certs from the certutil PKI secure every
connection; peers without a CA-signed cert are rejected at handshake
(reference security model, rfc/2025-05-30_mtls.md)."""

import subprocess
import sys
import time
from pathlib import Path

import pytest

core = pytest.importorskip("hypha_amd._core")
REPO = Path(__file__).resolve().parent.parent
TOOL = REPO / "tools" / "hypha_certutil.py"


@pytest.fixture(scope="module")
def pki(tmp_path_factory):
    out = tmp_path_factory.mktemp("pki")
    run = lambda *a: subprocess.run([sys.executable, str(TOOL), *a], check=True)
    run("root", "--out", str(out))
    run("org", "--out", str(out), "--name", "org1")
    for n in ("gw", "alice", "bob"):
        run("node", "--out", str(out), "--org", "org1", "--name", n)
    # a second, UNRELATED pki for the impostor
    evil = tmp_path_factory.mktemp("evil")
    run("root", "--out", str(evil))
    run("org", "--out", str(evil), "--name", "org1")
    run("node", "--out", str(evil), "--org", "org1", "--name", "mallory")
    return out, evil


def _tls_args(pki_dir, name):
    return dict(tls_cert=str(pki_dir / f"{name}.chain.pem"),
                tls_key=str(pki_dir / f"{name}.key"),
                tls_ca=str(pki_dir / "root.crt"))


def test_mtls_request_and_pubsub(pki):
    out, _ = pki
    gw = core.Gateway(str(out / "gw.chain.pem"), str(out / "gw.key"),
                      str(out / "root.crt"))
    gw.start(0)
    a = core.Node("alice", "127.0.0.1", gw.port, **_tls_args(out, "alice"))
    b = core.Node("bob", "127.0.0.1", gw.port, **_tls_args(out, "bob"))
    try:
        a.start(0)
        b.start(0)
        b.on("echo", lambda frm, body: {"x": body["x"] + 1})
        assert a.request("bob", "echo", {"x": 41}) == {"x": 42}
        got = []
        b.subscribe("t", lambda frm, d: got.append(d))
        time.sleep(0.05)
        a.publish("t", {"v": 9})
        time.sleep(0.2)
        assert got == [{"v": 9}]
    finally:
        a.stop()
        b.stop()
        gw.stop()


def test_mtls_rejects_foreign_ca(pki):
    out, evil = pki
    gw = core.Gateway(str(out / "gw.chain.pem"), str(out / "gw.key"),
                      str(out / "root.crt"))
    gw.start(0)
    mallory = core.Node("mallory", "127.0.0.1", gw.port, **_tls_args(evil, "mallory"))
    try:
        with pytest.raises(RuntimeError, match="TLS|registration"):
            mallory.start(0)
    finally:
        mallory.stop()
        gw.stop()


def test_crl_revoked_peer_rejected(pki, tmp_path_factory):
    """Revocation parity (rfc/2025-05-30_mtls.md CRL checking): after
    `certutil revoke`, a gateway loading the org CRL refuses the revoked
    node's handshake while still accepting unrevoked peers."""
    out = tmp_path_factory.mktemp("crlpki")
    run = lambda *a: subprocess.run([sys.executable, str(TOOL), *a], check=True)
    run("root", "--out", str(out))
    run("org", "--out", str(out), "--name", "org1")
    for n in ("gw", "carol", "dave"):
        run("node", "--out", str(out), "--org", "org1", "--name", n)
    run("crl", "--out", str(out), "--org", "org1")  # empty CRL first
    crl = str(out / "org1.crl.pem")

    gw = core.Gateway(str(out / "gw.chain.pem"), str(out / "gw.key"),
                      str(out / "root.crt"), crl)
    gw.start(0)
    carol = core.Node("carol", "127.0.0.1", gw.port, **_tls_args(out, "carol"))
    try:
        carol.start(0)  # accepted with an empty CRL
        carol.stop()
        run("revoke", "--out", str(out), "--org", "org1", "--name", "dave")
        gw2 = core.Gateway(str(out / "gw.chain.pem"), str(out / "gw.key"),
                           str(out / "root.crt"), crl)
        gw2.start(0)
        try:
            dave = core.Node("dave", "127.0.0.1", gw2.port, **_tls_args(out, "dave"))
            with pytest.raises(RuntimeError, match="TLS|registration"):
                dave.start(0)
            dave.stop()
            carol2 = core.Node("carol", "127.0.0.1", gw2.port,
                               **_tls_args(out, "carol"))
            carol2.start(0)  # unrevoked peer still accepted
            carol2.stop()
        finally:
            gw2.stop()
    finally:
        gw.stop()


def test_plaintext_client_rejected_by_tls_gateway(pki):
    out, _ = pki
    gw = core.Gateway(str(out / "gw.chain.pem"), str(out / "gw.key"),
                      str(out / "root.crt"))
    gw.start(0)
    plain = core.Node("plain", "127.0.0.1", gw.port)
    try:
        with pytest.raises(RuntimeError):
            plain.start(0)
    finally:
        plain.stop()
        gw.stop()


def _raw_tls_client(pki_dir, name, port):
    """Raw mTLS client speaking the 4-byte-BE-length + JSON wire framing, for
    crafting messages the Node class itself refuses to send (spoofed `from`,
    malformed frames)."""
    import json
    import socket
    import ssl
    import struct

    ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_CLIENT)
    ctx.check_hostname = False
    ctx.verify_mode = ssl.CERT_NONE  # client side; server still verifies US
    ctx.load_cert_chain(str(pki_dir / f"{name}.chain.pem"),
                        str(pki_dir / f"{name}.key"))
    raw = socket.create_connection(("127.0.0.1", port), timeout=5)
    s = ctx.wrap_socket(raw)

    def send(obj):
        data = json.dumps(obj).encode()
        s.sendall(struct.pack(">I", len(data)) + data)

    def recv():
        hdr = b""
        while len(hdr) < 4:
            chunk = s.recv(4 - len(hdr))
            if not chunk:
                return None
            hdr += chunk
        (n,) = struct.unpack(">I", hdr)
        body = b""
        while len(body) < n:
            chunk = s.recv(n - len(body))
            if not chunk:
                return None
            body += chunk
        return json.loads(body)

    return s, send, recv


def test_gateway_rejects_register_with_mismatched_cn(pki):
    """A peer holding alice's certificate cannot register as 'bob': the broker
    trusts only the verified CN (ADVICE: identity enforcement; reference
    model rfc/2025-05-30_mtls.md)."""
    out, _ = pki
    gw = core.Gateway(str(out / "gw.chain.pem"), str(out / "gw.key"),
                      str(out / "root.crt"))
    gw.start(0)
    try:
        s, send, recv = _raw_tls_client(out, "alice", gw.port)
        send({"kind": "register", "peer": "bob", "addr": "127.0.0.1:1"})
        resp = recv()
        assert resp is not None and resp.get("kind") == "error"
        s.close()
        # a truthful registration still works
        a = core.Node("alice", "127.0.0.1", gw.port, **_tls_args(out, "alice"))
        a.start(0)
        a.stop()
    finally:
        gw.stop()


def test_gateway_rejects_spoofed_from_on_request(pki):
    """kv_put sent with from='bob' over alice's certificate is refused, so an
    authenticated peer cannot impersonate another on the broker."""
    out, _ = pki
    gw = core.Gateway(str(out / "gw.chain.pem"), str(out / "gw.key"),
                      str(out / "root.crt"))
    gw.start(0)
    try:
        s, send, recv = _raw_tls_client(out, "alice", gw.port)
        send({"kind": "request", "type": "kv_put", "from": "bob",
              "body": {"key": "addr:bob", "value": "127.0.0.1:9"}})
        resp = recv()
        assert resp is not None and resp.get("ok") is False
        s.close()
    finally:
        gw.stop()


def test_node_rejects_spoofed_from(pki):
    """Node-to-node requests carry the verified CN: a 'from' field that
    contradicts the certificate is rejected before any handler runs, and
    handlers observe the CN, not the self-declared name."""
    import threading

    out, _ = pki
    gw = core.Gateway(str(out / "gw.chain.pem"), str(out / "gw.key"),
                      str(out / "root.crt"))
    gw.start(0)
    b = core.Node("bob", "127.0.0.1", gw.port, **_tls_args(out, "bob"))
    try:
        b.start(0)
        seen = []
        evt = threading.Event()

        def h(frm, body):
            seen.append(frm)
            evt.set()
            return {}

        b.on("probe", h)
        # direct raw connection to bob's listen port with alice's cert
        s, send, recv = _raw_tls_client(out, "alice", b.port)
        send({"kind": "request", "type": "probe", "from": "gw", "body": {}})
        resp = recv()
        assert resp is not None and resp.get("ok") is False
        assert "certificate" in resp.get("error", "")
        s.close()
        assert seen == []  # handler never ran for the spoofed message
        # and an honest request is attributed to the verified CN
        s2, send2, recv2 = _raw_tls_client(out, "alice", b.port)
        send2({"kind": "request", "type": "probe", "from": "alice", "body": {}})
        assert recv2().get("ok") is True
        s2.close()
        assert evt.wait(2) and seen == ["alice"]
    finally:
        b.stop()
        gw.stop()


def test_gateway_survives_malformed_messages(pki):
    """A register/subscribe/publish message missing required fields must drop
    only the offending connection, never the broker (ADVICE high: uncaught
    Json::at in a detached thread called std::terminate)."""
    out, _ = pki
    gw = core.Gateway(str(out / "gw.chain.pem"), str(out / "gw.key"),
                      str(out / "root.crt"))
    gw.start(0)
    try:
        # register with no 'peer'/'addr' fields -> Json::at throws server-side
        s, send, recv = _raw_tls_client(out, "alice", gw.port)
        send({"kind": "register"})
        assert recv() is None  # connection dropped, not the process
        s.close()
        # publish with no body -> handled inside try/catch as well
        s2, send2, recv2 = _raw_tls_client(out, "alice", gw.port)
        send2({"kind": "request", "type": "publish", "from": "alice",
               "body": {}})
        recv2()  # either an error response or a drop; gateway must survive
        s2.close()
        # broker still alive and serving
        a = core.Node("alice", "127.0.0.1", gw.port, **_tls_args(out, "alice"))
        a.start(0)
        a.kv_put("k", "v")
        assert a.kv_get("k") == "v"
        a.stop()
    finally:
        gw.stop()


def test_mtls_relay_circuit(pki):
    """End-to-end mTLS THROUGH a gateway relay circuit: when bob's direct
    address is unreachable, alice's connection is spliced by the gateway as
    opaque bytes — the peer TLS handshake still runs alice<->bob, so the
    relay cannot read or impersonate, and bob still sees the verified CN."""
    out, _ = pki
    gw = core.Gateway(str(out / "gw.chain.pem"), str(out / "gw.key"),
                      str(out / "root.crt"))
    gw.start(0)
    a = core.Node("alice", "127.0.0.1", gw.port, **_tls_args(out, "alice"))
    b = core.Node("bob", "127.0.0.1", gw.port, **_tls_args(out, "bob"))
    try:
        a.start(0)
        b.start(0)
        seen = []
        b.on("echo", lambda frm, body: (seen.append(frm), {"x": body["x"] + 1})[1])
        a.kv_put("addr:bob", "127.0.0.1:1")  # force the relay path
        assert a.request("bob", "echo", {"x": 41}) == {"x": 42}
        assert seen == ["alice"]  # identity from the VERIFIED cert CN
    finally:
        a.stop()
        b.stop()
        gw.stop()
