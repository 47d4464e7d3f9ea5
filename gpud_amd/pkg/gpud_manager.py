"""Package manager (reference: pkg/gpud-manager/controllers/
package_controller.go:19-341).

Watches ``<dataDir>/packages/*/init.sh`` package scripts and runs
reconcile loops: each package script supports the subcommands
``isInstalled`` / ``install`` / ``run`` / ``version`` (the reference's
contract); the controller converges every package to installed+running
and reports PackageStatus.
"""

from __future__ import annotations

import os
import threading
from typing import Dict, List, Optional

from ..apiv1.types import PackagePhase, PackageStatus
from .config import Config
from .log import logger
from .process_runner import run_bash


def discover_packages(packages_dir: str) -> Dict[str, str]:
    """package name -> init.sh path."""
    out: Dict[str, str] = {}
    if not os.path.isdir(packages_dir):
        return out
    for name in sorted(os.listdir(packages_dir)):
        init = os.path.join(packages_dir, name, "init.sh")
        if os.path.isfile(init):
            out[name] = init
    return out


def _run_pkg(init_sh: str, subcommand: str, timeout: float = 300.0):
    return run_bash(f'bash "{init_sh}" {subcommand}', timeout_seconds=timeout)


def package_status(name: str, init_sh: str) -> PackageStatus:
    st = PackageStatus(name=name, phase=PackagePhase.UNKNOWN)
    ver = _run_pkg(init_sh, "version", timeout=30)
    if ver.exit_code == 0:
        st.current_version = ver.output.strip().splitlines()[-1] if ver.output.strip() else ""
    installed = _run_pkg(init_sh, "isInstalled", timeout=60)
    if installed.exit_code == 0:
        st.phase = PackagePhase.INSTALLED
        st.status = "installed"
    else:
        st.phase = PackagePhase.INSTALLING
        st.status = "not installed"
    return st


def package_statuses(cfg: Config) -> List[PackageStatus]:
    return [
        package_status(name, init)
        for name, init in discover_packages(cfg.packages_dir).items()
    ]


class PackageController:
    """Reconcile loop converging every package to installed."""

    def __init__(self, cfg: Config, interval_seconds: float = 300.0):
        self.cfg = cfg
        self.interval = interval_seconds
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.statuses: Dict[str, PackageStatus] = {}
        self._lock = threading.Lock()

    def start(self) -> None:
        self._thread = threading.Thread(
            target=self._loop, daemon=True, name="gpud-pkg-manager"
        )
        self._thread.start()

    def reconcile_once(self) -> None:
        for name, init in discover_packages(self.cfg.packages_dir).items():
            st = package_status(name, init)
            if st.phase != PackagePhase.INSTALLED:
                logger.info("installing package %s", name)
                res = _run_pkg(init, "install")
                if res.exit_code == 0:
                    st = package_status(name, init)
                else:
                    st.status = f"install failed (exit {res.exit_code})"
            with self._lock:
                self.statuses[name] = st

    def _loop(self) -> None:
        self.reconcile_once()
        while not self._stop.wait(self.interval):
            try:
                self.reconcile_once()
            except Exception:
                logger.exception("package reconcile failed")

    def stop(self) -> None:
        self._stop.set()
