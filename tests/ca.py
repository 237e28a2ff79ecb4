"""Test CA: openssl-generated mutual-TLS material (reference
test/setup-ca.sh, which used certstrap).

Layout per CA directory: ca.crt / ca.key plus <name>.crt / <name>.key
for each component, CN = SAN = <name> (e.g. component.registry,
controller.host-0, host.host-0, user.admin).
"""

from __future__ import annotations

import os
import subprocess

DEFAULT_NAMES = (
    "component.registry",
    "controller.host-0",
    "host.host-0",
    "user.admin",
)


def _run(*cmd: str) -> None:
    subprocess.run(cmd, check=True, capture_output=True)


def make_ca(directory: str, names=DEFAULT_NAMES, ca_name: str = "OIM Test CA") -> str:
    os.makedirs(directory, exist_ok=True)
    ca_key = os.path.join(directory, "ca.key")
    ca_crt = os.path.join(directory, "ca.crt")
    _run(
        "openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
        "-keyout", ca_key, "-out", ca_crt, "-days", "2",
        "-subj", f"/CN={ca_name}",
    )
    for name in names:
        issue_cert(directory, name)
    return directory


def issue_cert(directory: str, name: str) -> None:
    key = os.path.join(directory, f"{name}.key")
    csr = os.path.join(directory, f"{name}.csr")
    crt = os.path.join(directory, f"{name}.crt")
    ext = os.path.join(directory, f"{name}.ext")
    _run(
        "openssl", "req", "-newkey", "rsa:2048", "-nodes",
        "-keyout", key, "-out", csr, "-subj", f"/CN={name}",
    )
    with open(ext, "w") as f:
        f.write(f"subjectAltName=DNS:{name}\n")
    _run(
        "openssl", "x509", "-req", "-in", csr,
        "-CA", os.path.join(directory, "ca.crt"),
        "-CAkey", os.path.join(directory, "ca.key"),
        "-CAcreateserial", "-out", crt, "-days", "2", "-extfile", ext,
    )
    os.unlink(csr)
    os.unlink(ext)


def tls_config(directory: str, name: str, peer_name: str = ""):
    from oim_amd.common import TLSConfig

    return TLSConfig(
        ca=os.path.join(directory, "ca.crt"),
        key=os.path.join(directory, f"{name}.key"),
        cert=os.path.join(directory, f"{name}.crt"),
        peer_name=peer_name,
    )
