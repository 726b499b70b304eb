"""Generate self-signed test certificates via the openssl CLI.

Parity: /root/reference/tool/generate_tls_certs.py (which uses the
``cryptography`` package — not available in this image, so we shell out to
openssl).  Produces a CA plus a server cert valid for localhost/127.0.0.1
under the given directory (default /tmp/rayfed_amd/test-certs/).
"""
from __future__ import annotations

import os
import subprocess
import sys
import tempfile


def generate(cert_dir: str = "/tmp/rayfed_amd/test-certs") -> dict:
    os.makedirs(cert_dir, exist_ok=True)
    ca_key = os.path.join(cert_dir, "ca.key")
    ca_crt = os.path.join(cert_dir, "ca.crt")
    srv_key = os.path.join(cert_dir, "server.key")
    srv_csr = os.path.join(cert_dir, "server.csr")
    srv_crt = os.path.join(cert_dir, "server.crt")

    def run(cmd):
        subprocess.run(cmd, check=True, capture_output=True)

    run([
        "openssl", "req", "-x509", "-newkey", "rsa:2048", "-keyout", ca_key,
        "-out", ca_crt, "-days", "365", "-nodes", "-subj", "/CN=rayfed-amd-test-ca",
    ])
    run([
        "openssl", "req", "-newkey", "rsa:2048", "-keyout", srv_key,
        "-out", srv_csr, "-nodes", "-subj", "/CN=localhost",
    ])
    with tempfile.NamedTemporaryFile("w", suffix=".ext", delete=False) as f:
        f.write("subjectAltName=DNS:localhost,IP:127.0.0.1\n")
        ext_file = f.name
    try:
        run([
            "openssl", "x509", "-req", "-in", srv_csr, "-CA", ca_crt,
            "-CAkey", ca_key, "-CAcreateserial", "-out", srv_crt,
            "-days", "365", "-extfile", ext_file,
        ])
    finally:
        os.unlink(ext_file)

    return {"ca_cert": ca_crt, "key": srv_key, "cert": srv_crt}


if __name__ == "__main__":
    out_dir = sys.argv[1] if len(sys.argv) > 1 else "/tmp/rayfed_amd/test-certs"
    paths = generate(out_dir)
    print(paths)
