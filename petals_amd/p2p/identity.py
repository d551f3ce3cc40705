"""Node identity: an ed25519 keypair + self-signed certificate whose SHA-256
fingerprint IS the node's peer id.

This binds the peer id gossiped through the DHT to a keypair the node proves
ownership of during the transport's STARTTLS handshake (the reference got
this from libp2p's ed25519 peer identities, reference server/server.py:92).
Certificates are generated with the system openssl (no python `cryptography`
package in the image); TLS itself is the stdlib `ssl` module.

Trust model: clients learn (peer_id -> address) from the DHT and verify at
connect time that the TLS certificate's fingerprint matches the announced
peer id — a relay or on-path attacker cannot impersonate a server without
its key. DHT records themselves are not signed (as in the reference's DHT,
subkey records are written by the announcing peer); a malicious DHT node can
still censor records.
"""

from __future__ import annotations

import hashlib
import os
import ssl
import subprocess
import tempfile
from typing import Optional

PEER_ID_BYTES = 16  # hex-encoded to 32 chars, matching the plaintext ids


class NodeIdentity:
    """Keypair + certificate on disk; peer_id = sha256(cert DER)[:16].hex()."""

    def __init__(self, identity_dir: Optional[str] = None):
        self._own_dir = identity_dir is None
        self.dir = identity_dir or tempfile.mkdtemp(prefix="petals-amd-id-")
        os.makedirs(self.dir, exist_ok=True)
        self.key_path = os.path.join(self.dir, "node_key.pem")
        self.cert_path = os.path.join(self.dir, "node_cert.pem")
        if not (os.path.exists(self.key_path) and os.path.exists(self.cert_path)):
            self._generate()
        self.cert_der = self._cert_der()
        self.peer_id = cert_fingerprint(self.cert_der)

    def _generate(self) -> None:
        res = subprocess.run(
            [
                "openssl", "req", "-x509", "-newkey", "ed25519",
                "-keyout", self.key_path, "-out", self.cert_path,
                "-nodes", "-days", "36500", "-subj", "/CN=petals-amd-node",
            ],
            capture_output=True,
            text=True,
        )
        if res.returncode != 0:
            raise RuntimeError(f"openssl certificate generation failed: {res.stderr[-500:]}")
        os.chmod(self.key_path, 0o600)

    def _cert_der(self) -> bytes:
        res = subprocess.run(
            ["openssl", "x509", "-in", self.cert_path, "-outform", "DER"],
            capture_output=True,
        )
        if res.returncode != 0:
            raise RuntimeError("openssl cert DER conversion failed")
        return res.stdout

    def server_ssl_context(self) -> ssl.SSLContext:
        ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
        ctx.load_cert_chain(self.cert_path, self.key_path)
        ctx.minimum_version = ssl.TLSVersion.TLSv1_3
        return ctx

    @staticmethod
    def client_ssl_context() -> ssl.SSLContext:
        # self-signed swarm certs: authenticity comes from the fingerprint ==
        # announced-peer-id check after the handshake, not from a CA
        ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_CLIENT)
        ctx.check_hostname = False
        ctx.verify_mode = ssl.CERT_NONE
        ctx.minimum_version = ssl.TLSVersion.TLSv1_3
        return ctx


def cert_fingerprint(cert_der: bytes) -> str:
    return hashlib.sha256(cert_der).digest()[:PEER_ID_BYTES].hex()
