#!/usr/bin/env python3
"""Development PKI generator — 3-tier Ed25519 chain (Root CA -> Org CA ->
node certs) with trust-chain bundles.

Parity with the reference's certutil crate
(/root/reference/crates/certutil: subcommands root/org/node,
docs/quickstart.md:48-58 usage). Uses the system openssl CLI. The control
plane's mTLS transport wiring is tracked for round 2; these certificates
carry the same identity model (PeerId derived from the node key).

Usage:
  python tools/hypha_certutil.py root --out pki/
  python tools/hypha_certutil.py org --out pki/ --name org1
  python tools/hypha_certutil.py node --out pki/ --org org1 --name worker-0
  python tools/hypha_certutil.py crl --out pki/ --org org1        # (re)issue CRL
  python tools/hypha_certutil.py revoke --out pki/ --org org1 --name worker-0

Revocation parity: the reference's forked libp2p mTLS checks CRLs
(rfc/2025-05-30_mtls.md; network/src/cert.rs load_crls_from_pem). `revoke`
adds a node cert to the org's revocation list and re-issues
`<org>.crl.pem`, which the daemons load via --tls-crl.
"""

from __future__ import annotations

import argparse
import hashlib
import subprocess
import sys
from pathlib import Path


def run(*cmd):
    subprocess.run([str(c) for c in cmd], check=True, capture_output=True)


def gen_key(path: Path):
    run("openssl", "genpkey", "-algorithm", "ed25519", "-out", path)


def self_signed_ca(key: Path, crt: Path, cn: str):
    run("openssl", "req", "-new", "-x509", "-key", key, "-out", crt, "-days", "3650",
        "-subj", f"/CN={cn}", "-addext", "basicConstraints=critical,CA:TRUE")


def signed_cert(key: Path, ca_key: Path, ca_crt: Path, crt: Path, cn: str, ca: bool):
    csr = crt.with_suffix(".csr")
    run("openssl", "req", "-new", "-key", key, "-out", csr, "-subj", f"/CN={cn}")
    ext = crt.with_suffix(".ext")
    ext.write_text(
        "basicConstraints=critical,CA:TRUE\n" if ca
        else "basicConstraints=critical,CA:FALSE\nsubjectAltName=DNS:localhost,IP:127.0.0.1\n"
    )
    run("openssl", "x509", "-req", "-in", csr, "-CA", ca_crt, "-CAkey", ca_key,
        "-CAcreateserial", "-out", crt, "-days", "1825", "-extfile", ext)
    csr.unlink()
    ext.unlink()


def _cert_field(crt: Path, flag: str) -> str:
    out = subprocess.run(
        ["openssl", "x509", "-in", str(crt), flag, "-noout"],
        check=True, capture_output=True, text=True,
    ).stdout.strip()
    return out.split("=", 1)[1] if "=" in out else out


def _index_date(openssl_date: str) -> str:
    """'May  3 12:00:00 2031 GMT' -> openssl-ca index format YYMMDDHHMMSSZ."""
    from datetime import datetime

    dt = datetime.strptime(" ".join(openssl_date.split()), "%b %d %H:%M:%S %Y %Z")
    return dt.strftime("%y%m%d%H%M%S") + "Z"


def issue_crl(out: Path, org: str) -> Path:
    """(Re)issue the org CA's CRL from its revocation list `<org>.revoked`
    (tab-separated openssl-ca index rows appended by `revoke`)."""
    index = out / f"{org}.revoked"
    index.touch()
    # openssl ca requires the .attr sidecar and a crlnumber counter
    (out / f"{org}.revoked.attr").write_text("unique_subject = no\n")
    crlnum = out / f"{org}.crlnumber"
    if not crlnum.exists():
        crlnum.write_text("01\n")
    cfg = out / f"{org}.ca.cnf"
    cfg.write_text(
        "[ca]\ndefault_ca = CA_default\n[CA_default]\n"
        f"database = {index}\ncrlnumber = {crlnum}\ndefault_md = default\n"
    )
    crl = out / f"{org}.crl.pem"
    run("openssl", "ca", "-config", cfg, "-gencrl", "-keyfile", out / f"{org}.key",
        "-cert", out / f"{org}.crt", "-out", crl, "-crldays", "3650")
    return crl


def revoke_cert(out: Path, org: str, name: str) -> Path:
    """Append the node cert to the org's revocation index and re-issue CRL."""
    from datetime import datetime, timezone

    crt = out / f"{name}.crt"
    serial = _cert_field(crt, "-serial")
    expiry = _index_date(_cert_field(crt, "-enddate"))
    now = datetime.now(timezone.utc).strftime("%y%m%d%H%M%S") + "Z"
    subject = _cert_field(crt, "-subject").strip()
    if not subject.startswith("/"):
        subject = "/" + subject.replace(" = ", "=").replace(", ", "/")
    index = out / f"{org}.revoked"
    existing = index.read_text() if index.exists() else ""
    if f"\t{serial}\t" not in existing:
        with index.open("a") as f:
            f.write(f"R\t{expiry}\t{now}\t{serial}\tunknown\t{subject}\n")
    return issue_crl(out, org)


def peer_id(crt: Path) -> str:
    """PeerId = hash of the certificate's public key (the reference derives
    libp2p PeerIds from the cert key, crates/network/src/cert.rs:30)."""
    out = subprocess.run(
        ["openssl", "x509", "-in", str(crt), "-pubkey", "-noout"],
        check=True, capture_output=True, text=True,
    ).stdout
    return "peer-" + hashlib.sha256(out.encode()).hexdigest()[:16]


def main() -> int:
    p = argparse.ArgumentParser()
    sub = p.add_subparsers(dest="cmd", required=True)
    for name in ("root", "org", "node", "crl", "revoke"):
        sp = sub.add_parser(name)
        sp.add_argument("--out", default="pki")
        if name in ("org", "node", "revoke"):
            sp.add_argument("--name", required=True)
        if name in ("node", "crl", "revoke"):
            sp.add_argument("--org", required=True)
    args = p.parse_args()
    out = Path(args.out)
    out.mkdir(parents=True, exist_ok=True)

    if args.cmd == "root":
        gen_key(out / "root.key")
        self_signed_ca(out / "root.key", out / "root.crt", "hypha-root")
        print(f"root CA: {out/'root.crt'}")
    elif args.cmd == "org":
        gen_key(out / f"{args.name}.key")
        signed_cert(out / f"{args.name}.key", out / "root.key", out / "root.crt",
                    out / f"{args.name}.crt", f"hypha-org-{args.name}", ca=True)
        print(f"org CA: {out/(args.name + '.crt')}")
    elif args.cmd == "node":
        key = out / f"{args.name}.key"
        crt = out / f"{args.name}.crt"
        gen_key(key)
        signed_cert(key, out / f"{args.org}.key", out / f"{args.org}.crt", crt,
                    args.name, ca=False)
        # trust-chain bundle: node + org + root
        chain = out / f"{args.name}.chain.pem"
        chain.write_text(crt.read_text() + (out / f"{args.org}.crt").read_text()
                         + (out / "root.crt").read_text())
        print(f"node cert: {crt}  peer-id: {peer_id(crt)}")
    elif args.cmd == "crl":
        print(f"crl: {issue_crl(out, args.org)}")
    elif args.cmd == "revoke":
        print(f"revoked {args.name}; crl: {revoke_cert(out, args.org, args.name)}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
