"""Support-bundle collector — the nvidia-bug-report.sh analog.

Reference: pkg/session/diagnostic.go:27-183 runs `nvidia-bug-report.sh`,
gzips the output and uploads it to a presigned URL. There is no vendor
bug-report script to exec on an MI355X node, so the bundle is collected
natively: live amdsmi state (snapshots, static info, partition mode, bad
pages), the kernel-message tail, daemon state (component health, config,
metadata keys), and host context (kernel, modules, PCI topology) into one
tar.gz a support engineer can open anywhere.
"""

from __future__ import annotations

import io
import json
import os
import subprocess
import tarfile
import time
from typing import Any, Callable, Dict, List, Optional

from .. import __version__
from .log import logger

# each entry: (archive name, collector returning str) — collectors must not
# raise; failures become "<name>.error" members so a partial bundle is
# still useful
Collector = Callable[[], str]


def _json(obj: Any) -> str:
    return json.dumps(obj, indent=1, default=str, sort_keys=True)


def _run_cmd(argv: List[str], timeout: float = 15.0) -> str:
    out = subprocess.run(argv, capture_output=True, text=True, timeout=timeout)
    return out.stdout + (("\n[stderr]\n" + out.stderr) if out.stderr else "")


def _smi_dump(smi) -> str:
    data: Dict[str, Any] = {
        "product": smi.product_name,
        "driver_version": smi.driver_version,
        "rocm_version": smi.rocm_version,
        "device_count": smi.device_count(),
        "snapshots": smi.snapshot_all(),
        "devices": {},
    }
    for uuid, dev in smi.devices().items():
        d: Dict[str, Any] = {}
        for attr in ("vram_info", "vbios_info", "partition_info", "bad_page_info"):
            try:
                d[attr] = getattr(dev, attr)()
            except Exception as e:  # noqa: BLE001 — partial dumps are fine
                d[attr] = {"error": str(e)}
        data["devices"][uuid] = d
    return _json(data)


def _kmsg_tail(max_lines: int = 2000) -> str:
    """Ring-buffer tail via a non-blocking /dev/kmsg read (works without
    the dmesg binary; SYSLOG_ACTION_READ_ALL needs CAP_SYSLOG anyway)."""
    lines: List[str] = []
    errors = 0
    fd = os.open("/dev/kmsg", os.O_RDONLY | os.O_NONBLOCK)
    try:
        while len(lines) < 20000 and errors < 10000:
            try:
                chunk = os.read(fd, 8192)
            except BlockingIOError:
                break
            except OSError:
                errors += 1  # EPIPE: writer overtook us — skip the lost record
                continue
            if not chunk:
                break
            lines.append(chunk.decode("utf-8", "replace").rstrip("\n"))
    finally:
        os.close(fd)
    return "\n".join(lines[-max_lines:])


def _states_dump(core) -> str:
    states = {}
    for comp in core.registry.all_components():
        try:
            crs = comp.last_health_states()
            states[comp.name] = [
                h.to_dict() if hasattr(h, "to_dict") else h for h in crs
            ]
        except Exception as e:  # noqa: BLE001
            states[comp.name] = [{"error": str(e)}]
    return _json(states)


def collect_bundle(
    out_path: str,
    core: Any = None,
    smi: Any = None,
    extra_files: Optional[Dict[str, str]] = None,
) -> str:
    """Write the tar.gz bundle and return its path."""
    smi = smi or (core.smi_instance if core is not None else None)
    collectors: List[tuple] = [
        (
            "bundle-info.json",
            lambda: _json(
                {
                    "gpud_version": __version__,
                    "collected_at": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
                    "hostname": os.uname().nodename,
                    "kernel": " ".join(os.uname()),
                }
            ),
        ),
        ("kmsg.log", _kmsg_tail),
        ("lspci.txt", lambda: _run_cmd(["lspci", "-vvv"], timeout=30)),
        ("modules.txt", lambda: open("/proc/modules").read()),
        ("meminfo.txt", lambda: open("/proc/meminfo").read()),
        ("cmdline.txt", lambda: open("/proc/cmdline").read()),
    ]
    if smi is not None and getattr(smi, "exists", False):
        collectors.append(("amdsmi.json", lambda: _smi_dump(smi)))
    if core is not None:
        collectors.append(("states.json", lambda: _states_dump(core)))
        collectors.append(
            ("config.json", lambda: _json(getattr(core, "config").__dict__))
        )
    os.makedirs(os.path.dirname(os.path.abspath(out_path)), exist_ok=True)
    with tarfile.open(out_path, "w:gz") as tf:
        for name, fn in collectors:
            try:
                content = fn()
            except Exception as e:  # noqa: BLE001 — keep collecting
                name, content = name + ".error", str(e)
            data = content.encode("utf-8", "replace")
            info = tarfile.TarInfo(name=f"gpud-bundle/{name}")
            info.size = len(data)
            info.mtime = int(time.time())
            tf.addfile(info, io.BytesIO(data))
        for name, content in (extra_files or {}).items():
            data = content.encode("utf-8", "replace")
            info = tarfile.TarInfo(name=f"gpud-bundle/{name}")
            info.size = len(data)
            info.mtime = int(time.time())
            tf.addfile(info, io.BytesIO(data))
    logger.info("support bundle written: %s (%d bytes)",
                out_path, os.path.getsize(out_path))
    return out_path


def upload_bundle(path: str, url: str, timeout: float = 120.0) -> Optional[str]:
    """PUT the bundle to a presigned URL (reference: diagnostic.go upload).
    Returns an error string or None."""
    import httpx

    try:
        with open(path, "rb") as f:
            r = httpx.put(
                url,
                content=f.read(),
                headers={"Content-Type": "application/gzip"},
                timeout=timeout,
            )
        if r.status_code >= 300:
            return f"upload failed: HTTP {r.status_code}"
        return None
    except httpx.HTTPError as e:
        return f"upload failed: {e}"
