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
    for name in ("root", "org", "node"):
        sp = sub.add_parser(name)
        sp.add_argument("--out", default="pki")
        if name != "root":
            sp.add_argument("--name", required=True)
        if name == "node":
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
    return 0


if __name__ == "__main__":
    sys.exit(main())
