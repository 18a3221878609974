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
