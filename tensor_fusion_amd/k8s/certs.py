"""Self-signed TLS material for the admission webhook.

A real apiserver only calls webhooks over TLS with a caBundle it trusts;
the reference's Helm chart provisions this via cert-manager or a
generator job. This module shells out to openssl (present in the image)
to produce a CA + server certificate for the operator's webhook service
and returns the base64 caBundle to embed in the
MutatingWebhookConfiguration.

    python -m tensor_fusion_amd.k8s.certs /path/outdir tf-operator.tf-system.svc
"""
from __future__ import annotations

import base64
import os
import subprocess
import sys
from typing import List, Optional


def generate(outdir: str, cn: str,
             sans: Optional[List[str]] = None) -> str:
    """Write ca.crt, tls.crt, tls.key into outdir; returns b64 caBundle."""

    os.makedirs(outdir, exist_ok=True)
    sans = sans or [cn, "localhost", "127.0.0.1"]
    ca_key = os.path.join(outdir, "ca.key")
    ca_crt = os.path.join(outdir, "ca.crt")
    key = os.path.join(outdir, "tls.key")
    csr = os.path.join(outdir, "tls.csr")
    crt = os.path.join(outdir, "tls.crt")
    ext = os.path.join(outdir, "san.cnf")

    def run(*cmd: str):
        subprocess.run(cmd, check=True, capture_output=True)

    run("openssl", "genrsa", "-out", ca_key, "2048")
    run("openssl", "req", "-x509", "-new", "-nodes", "-key", ca_key,
        "-subj", "/CN=tensor-fusion-ca", "-days", "3650", "-out", ca_crt)
    run("openssl", "genrsa", "-out", key, "2048")
    run("openssl", "req", "-new", "-key", key, "-subj", f"/CN={cn}",
        "-out", csr)
    alt = ",".join(
        f"IP:{s}" if s.replace(".", "").isdigit() else f"DNS:{s}"
        for s in sans)
    with open(ext, "w") as f:
        f.write(f"subjectAltName={alt}\n")
    run("openssl", "x509", "-req", "-in", csr, "-CA", ca_crt,
        "-CAkey", ca_key, "-CAcreateserial", "-days", "3650",
        "-extfile", ext, "-out", crt)
    with open(ca_crt, "rb") as f:
        return base64.b64encode(f.read()).decode()


if __name__ == "__main__":
    outdir = sys.argv[1]
    cn = sys.argv[2] if len(sys.argv) > 2 else \
        "tensor-fusion-operator.tf-system.svc"
    print(generate(outdir, cn))
