"""systemd install/uninstall (reference: pkg/gpud-manager/systemd/ — unit
at /etc/systemd/system/gpud.service, env file /etc/default/gpud,
Restart=always / RestartSec=5s — gpud.service:11,21-22)."""

from __future__ import annotations

import os
import shutil
import subprocess
import sys
from typing import Optional

UNIT_PATH = "/etc/systemd/system/gpud-amd.service"
ENV_PATH = "/etc/default/gpud-amd"

UNIT_TEMPLATE = """[Unit]
Description=gpud-amd (MI355X GPU health daemon)
After=network.target

[Service]
Type=simple
EnvironmentFile=-{env_path}
ExecStart={python} -m gpud_amd run --data-dir {data_dir} $FLAGS
Restart=always
RestartSec=5s
LimitNOFILE=65536

[Install]
WantedBy=multi-user.target
"""


def _systemctl(*args: str) -> Optional[str]:
    if shutil.which("systemctl") is None:
        return "systemctl not available on this host"
    try:
        out = subprocess.run(
            ["systemctl", *args], capture_output=True, text=True, timeout=30
        )
        if out.returncode != 0:
            return f"systemctl {' '.join(args)} failed: {out.stderr.strip()}"
    except (OSError, subprocess.TimeoutExpired) as e:
        return f"systemctl {' '.join(args)} failed: {e}"
    return None


def install_and_start(
    data_dir: str, token: str = "", endpoint: str = ""
) -> Optional[str]:
    flags = []
    if endpoint:
        flags += ["--endpoint", endpoint]
    if token:
        flags += ["--token", token]
    try:
        os.makedirs(data_dir, exist_ok=True)
        with open(ENV_PATH, "w") as f:
            f.write(f"FLAGS={' '.join(flags)}\n")
        with open(UNIT_PATH, "w") as f:
            f.write(
                UNIT_TEMPLATE.format(
                    env_path=ENV_PATH,
                    python=sys.executable,
                    data_dir=data_dir,
                )
            )
    except OSError as e:
        return f"cannot write systemd unit: {e}"
    for args in (["daemon-reload"], ["enable", "gpud-amd"], ["restart", "gpud-amd"]):
        err = _systemctl(*args)
        if err:
            return err
    return None


def stop_and_disable() -> Optional[str]:
    for args in (["stop", "gpud-amd"], ["disable", "gpud-amd"]):
        err = _systemctl(*args)
        if err:
            return err
    return None
