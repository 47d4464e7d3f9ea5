"""kap-mTLS credential manager (reference: pkg/kapmtls/manager.go:105-245).

Stages control-plane-issued client certificates for a node-local agent and
activates them atomically (symlink swap) with rollback: the control plane
pushes new credentials via the session (updateKAPMTLSCredentials), then
asks for activation (activateKAPMTLS); a failed activation restores the
previous credentials.
"""

from __future__ import annotations

import os
import re
import time
from typing import Dict, Optional

ACTIVE_LINK = "active"
STAGED_DIR = "staged"
PREVIOUS_DIR = "previous"

# The version string is control-plane-supplied and becomes a path segment;
# confine it to a safe charset so 'v-{version}' can never traverse out of
# base_dir (no separators, no traversal, no hidden-file tricks).
_VERSION_RE = re.compile(r"^[A-Za-z0-9][A-Za-z0-9._-]{0,127}$")


def _valid_version(version: str) -> bool:
    return bool(_VERSION_RE.match(version)) and ".." not in version


class Manager:
    def __init__(self, base_dir: str):
        self.base_dir = base_dir
        os.makedirs(base_dir, exist_ok=True)

    # -- paths --------------------------------------------------------------

    @property
    def active_link(self) -> str:
        return os.path.join(self.base_dir, ACTIVE_LINK)

    def _versioned_dir(self, version: str) -> str:
        return os.path.join(self.base_dir, f"v-{version}")

    # -- operations ----------------------------------------------------------

    def stage(
        self,
        cert_pem: bytes,
        key_pem: bytes,
        version: str = "",
        gateway_ca_pem: bytes = b"",
        gateway_endpoint: str = "",
        server_name: str = "",
    ) -> str:
        """Write new credentials into a staged versioned directory. The
        optional gateway metadata (CA bundle, endpoint, TLS server name —
        reference: UpdateKAPMTLSCredentialsRequest fields) travels with
        the version so rollback restores the matching gateway config."""
        version = version or str(int(time.time() * 1000))
        if not _valid_version(version):
            raise ValueError(f"invalid kap-mTLS credential version {version!r}")
        d = self._versioned_dir(version)
        os.makedirs(d, exist_ok=True)
        with open(os.path.join(d, "client.crt"), "wb") as f:
            f.write(cert_pem)
        key_path = os.path.join(d, "client.key")
        with open(key_path, "wb") as f:
            f.write(key_pem)
        os.chmod(key_path, 0o600)
        if gateway_ca_pem:
            with open(os.path.join(d, "gateway-ca.crt"), "wb") as f:
                f.write(gateway_ca_pem)
        if gateway_endpoint or server_name:
            import json as _json

            with open(os.path.join(d, "gateway.json"), "w") as f:
                _json.dump(
                    {"endpoint": gateway_endpoint, "server_name": server_name},
                    f,
                )
        with open(os.path.join(self.base_dir, STAGED_DIR), "w") as f:
            f.write(version)
        return version

    def gateway_info(self) -> Dict[str, str]:
        """Gateway metadata of the ACTIVE version ({} when absent)."""
        try:
            import json as _json

            with open(os.path.join(self.active_link, "gateway.json")) as f:
                return _json.load(f)
        except (OSError, ValueError):
            return {}

    def staged_version(self) -> str:
        try:
            with open(os.path.join(self.base_dir, STAGED_DIR)) as f:
                return f.read().strip()
        except OSError:
            return ""

    def active_version(self) -> str:
        try:
            target = os.readlink(self.active_link)
            return os.path.basename(target).removeprefix("v-")
        except OSError:
            return ""

    def activate(self, version: str = "") -> Optional[str]:
        """Atomically point ``active`` at the staged (or given) version.

        Returns an error string, or None. The previous target is remembered
        for rollback.
        """
        version = version or self.staged_version()
        if not version:
            return "no staged credentials to activate"
        if not _valid_version(version):
            return f"invalid kap-mTLS credential version {version!r}"
        d = self._versioned_dir(version)
        if not os.path.isdir(d):
            return f"staged credential dir missing: {d}"
        for req in ("client.crt", "client.key"):
            if not os.path.exists(os.path.join(d, req)):
                return f"staged credentials incomplete: {req} missing"
        prev = self.active_version()
        if prev:
            with open(os.path.join(self.base_dir, PREVIOUS_DIR), "w") as f:
                f.write(prev)
        tmp_link = self.active_link + ".tmp"
        try:
            if os.path.lexists(tmp_link):
                os.unlink(tmp_link)
            os.symlink(d, tmp_link)
            os.replace(tmp_link, self.active_link)  # atomic swap
        except OSError as e:
            return f"activation failed: {e}"
        return None

    def rollback(self) -> Optional[str]:
        try:
            with open(os.path.join(self.base_dir, PREVIOUS_DIR)) as f:
                prev = f.read().strip()
        except OSError:
            return "no previous credentials recorded"
        return self.activate(prev)

    def active_paths(self) -> Optional[Dict[str, str]]:
        """Cert/key paths of the ACTIVE credentials, or None."""
        v = self.active_version()
        if not v:
            return None
        d = self._versioned_dir(v)
        return {
            "cert": os.path.join(d, "client.crt"),
            "key": os.path.join(d, "client.key"),
        }

    def grpc_channel_credentials(self, root_ca_pem: Optional[bytes] = None):
        """Build gRPC mTLS channel credentials from the active credentials
        (feeds the v2 session — reference: kap-mTLS staged client certs for
        the node-local agent)."""
        import grpc

        paths = self.active_paths()
        if paths is None:
            return None
        with open(paths["cert"], "rb") as f:
            cert = f.read()
        with open(paths["key"], "rb") as f:
            key = f.read()
        return grpc.ssl_channel_credentials(
            root_certificates=root_ca_pem,
            private_key=key,
            certificate_chain=cert,
        )

    def status(self) -> Dict[str, str]:
        active = self.active_version()
        st = {
            "active_version": active,
            "staged_version": self.staged_version(),
        }
        if active:
            d = self._versioned_dir(active)
            st["active_cert"] = os.path.join(d, "client.crt")
            st["active_key"] = os.path.join(d, "client.key")
        return st
