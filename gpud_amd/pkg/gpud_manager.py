"""Package manager (reference: pkg/gpud-manager).

Watches ``<dataDir>/packages/*/init.sh`` package scripts and converges each
package through the reference's full controller set
(controllers/package_controller.go:46-341):

  * **install runner** — dependency-gated: a package installs only after
    every ``#GPUD_PACKAGE_DEPENDENCY`` entry (``name:minversion`` or
    ``name:*``) is installed at a sufficient version; ``shouldSkip`` (exit
    0) marks the package Skipped; otherwise ``isInstalled`` is probed and a
    failed probe triggers ``install`` then ``start``.
  * **update runner** — for installed packages, reads ``version`` and
    compares against the ``#GPUD_PACKAGE_VERSION`` target from the script
    header; a mismatch triggers ``upgrade`` with progress estimated from
    ``#GPUD_PACKAGE_INSTALL_TIME`` (reference: the 2-second progress
    ticker, capped at 98 until done).
  * **status runner** — for installed packages, ``status`` (exit 0 = ok);
    a failing status triggers ``stop`` then ``start`` (restart).
  * **delete runner** — ``needDelete`` exit 0 triggers ``delete``.

Each subcommand's output is logged to ``<pkgdir>/<subcommand>.log``
(reference: runCommand's per-arg log files). The file informer is a
polling rescan (3 s, same cadence as the reference's syncPeriod) instead
of inotify — re-resolving a package's metadata whenever init.sh's mtime
changes.

Script contract (subcommands a package's init.sh must answer):
``isInstalled install start stop status version upgrade shouldSkip
needDelete delete`` — plus header metadata lines::

    #GPUD_PACKAGE_VERSION=1.2.3
    #GPUD_PACKAGE_DEPENDENCY=other:1.0,base:*
    #GPUD_PACKAGE_INSTALL_TIME=5m
"""

from __future__ import annotations

import os
import re
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..apiv1.types import PackagePhase, PackageStatus
from .config import Config
from .log import logger
from .process_runner import RunResult, run_bash, stream_bash

SYNC_PERIOD_SECONDS = 3.0  # reference: package_controller.go syncPeriod

_DURATION_RE = re.compile(r"(\d+(?:\.\d+)?)\s*(h|m|s|ms)")


def parse_duration_seconds(raw: str) -> float:
    """Parse a Go-style duration ('5m', '1h30m', '90s') into seconds."""
    total = 0.0
    for num, unit in _DURATION_RE.findall(raw.strip()):
        total += float(num) * {"h": 3600.0, "m": 60.0, "s": 1.0, "ms": 0.001}[unit]
    return total


@dataclass
class PackageInfo:
    """Metadata resolved from a package's init.sh header (reference:
    informer/file_informer.go resolvePackage)."""

    name: str = ""
    script_path: str = ""
    target_version: str = ""
    dependency: List[List[str]] = field(default_factory=list)
    total_time_seconds: float = 0.0


@dataclass
class ManagedPackage:
    """Controller-side status (reference: packages.PackageStatus — the rich
    shape served at /admin/packages; the v1 API shape derives from it)."""

    name: str = ""
    script_path: str = ""
    target_version: str = ""
    current_version: str = ""
    dependency: List[List[str]] = field(default_factory=list)
    total_time_seconds: float = 0.0
    skipped: bool = False
    is_installed: bool = False
    installing: bool = False
    progress: int = 0
    status: bool = False
    install_started: Optional[float] = None

    def observed_progress(self) -> int:
        if not self.installing:
            return self.progress
        if not self.install_started or self.total_time_seconds <= 0:
            return self.progress
        pct = int((time.time() - self.install_started) / self.total_time_seconds * 100)
        return min(pct, 98)  # reference caps the estimate at 98

    def to_dict(self) -> Dict:
        return {
            "name": self.name,
            "skipped": self.skipped,
            "is_installed": self.is_installed,
            "installing": self.installing,
            "progress": self.observed_progress(),
            "total_time": int(self.total_time_seconds * 1e9),  # Go Duration ns
            "status": self.status,
            "target_version": self.target_version,
            "current_version": self.current_version,
            "script_path": self.script_path,
            "dependency": self.dependency,
        }

    def to_api(self) -> PackageStatus:
        if self.skipped:
            phase = PackagePhase.SKIPPED
        elif self.is_installed:
            phase = PackagePhase.INSTALLED
        elif self.installing:
            phase = PackagePhase.INSTALLING
        else:
            phase = PackagePhase.INSTALLING  # converging toward install
        return PackageStatus(
            name=self.name,
            phase=phase,
            status="installed" if self.is_installed else "not installed",
            current_version=self.current_version,
        )


def resolve_package(script_path: str) -> PackageInfo:
    """Parse the metadata header of an init.sh (reference:
    resolvePackage — grep #GPUD_PACKAGE_VERSION / _DEPENDENCY /
    _INSTALL_TIME, '=' separated)."""
    info = PackageInfo(
        name=os.path.basename(os.path.dirname(script_path)),
        script_path=script_path,
    )
    try:
        with open(script_path) as f:
            for line in f:
                line = line.strip()
                if line.startswith("#GPUD_PACKAGE_VERSION="):
                    info.target_version = line.split("=", 1)[1].strip()
                elif line.startswith("#GPUD_PACKAGE_DEPENDENCY="):
                    raw = line.split("=", 1)[1].strip()
                    for dep in raw.split(","):
                        parts = dep.split(":")
                        if len(parts) == 2:
                            info.dependency.append([parts[0].strip(), parts[1].strip()])
                elif line.startswith("#GPUD_PACKAGE_INSTALL_TIME="):
                    info.total_time_seconds = parse_duration_seconds(
                        line.split("=", 1)[1]
                    )
    except OSError as e:
        logger.warning("resolve package %s: %s", script_path, e)
    return info


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


def _run_pkg(init_sh: str, subcommand: str, timeout: float = 300.0,
             log_output: bool = True, stream: bool = False):
    """Run one init.sh subcommand; mirror the reference's per-subcommand
    log file (<pkgdir>/<arg>.log) so operators can inspect what each
    lifecycle step printed. ``stream=True`` (long installs/upgrades)
    writes the log LIVE line-by-line like the reference's streaming
    process reader, so a wedged install is inspectable mid-run."""
    if stream and log_output:
        log_path = os.path.join(os.path.dirname(init_sh), f"{subcommand}.log")
        lines = []
        timed_out = False
        holder = {}
        try:
            with open(log_path, "w") as f:
                try:
                    for line in stream_bash(
                        f'bash "{init_sh}" {subcommand}',
                        timeout_seconds=timeout,
                        result_holder=holder,
                    ):
                        lines.append(line)
                        f.write(line + "\n")
                        f.flush()
                except TimeoutError:
                    timed_out = True
        except OSError:
            # log dir unwritable: fall back to the buffered path
            return _run_pkg(init_sh, subcommand, timeout, log_output=False)
        return RunResult(
            exit_code=-1 if timed_out else holder.get("exit_code", -1),
            output="\n".join(lines),
            timed_out=timed_out,
            error="script timed out" if timed_out else "",
        )
    res = run_bash(f'bash "{init_sh}" {subcommand}', timeout_seconds=timeout)
    if log_output:
        try:
            log_path = os.path.join(os.path.dirname(init_sh), f"{subcommand}.log")
            with open(log_path, "w") as f:
                f.write(res.output)
        except OSError:
            pass
    return res


def package_status(name: str, init_sh: str) -> PackageStatus:
    st = PackageStatus(name=name, phase=PackagePhase.UNKNOWN)
    ver = _run_pkg(init_sh, "version", timeout=30, log_output=False)
    if ver.exit_code == 0:
        st.current_version = (
            ver.output.strip().splitlines()[-1] if ver.output.strip() else ""
        )
    skip = _run_pkg(init_sh, "shouldSkip", timeout=30, log_output=False)
    if skip.exit_code == 0:
        st.phase = "Skipped"
        st.status = "skipped"
        return st
    installed = _run_pkg(init_sh, "isInstalled", timeout=60, log_output=False)
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
    """Reconcile loops converging every package to installed + running at
    its target version (reference: package_controller.go Run — reconcile/
    install/update/status/delete runners)."""

    def __init__(self, cfg: Config, interval_seconds: float = SYNC_PERIOD_SECONDS):
        self.cfg = cfg
        self.interval = interval_seconds
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.packages: Dict[str, ManagedPackage] = {}
        self._mtimes: Dict[str, float] = {}
        self._lock = threading.Lock()
        # inline (synchronous) installs keep reconciliation deterministic;
        # set False to run install/upgrade in background threads (the
        # reference's goroutine shape) when a slow install must not block
        # the status runner
        self.inline_installs = True

    # -- back-compat view (older callers read .statuses of api shapes) -----
    @property
    def statuses(self) -> Dict[str, PackageStatus]:
        with self._lock:
            return {n: p.to_api() for n, p in self.packages.items()}

    def admin_statuses(self) -> List[Dict]:
        with self._lock:
            return [p.to_dict() for _, p in sorted(self.packages.items())]

    def start(self) -> None:
        self._thread = threading.Thread(
            target=self._loop, daemon=True, name="gpud-pkg-manager"
        )
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()

    def _loop(self) -> None:
        while not self._stop.is_set():
            try:
                self.reconcile_once()
            except Exception:
                logger.exception("package reconcile failed")
            if self._stop.wait(self.interval):
                return

    # -- informer (reference: informer/file_informer.go, polling variant) --

    def _informer_pass(self) -> None:
        found = discover_packages(self.cfg.packages_dir)
        for name, init in found.items():
            try:
                mtime = os.stat(init).st_mtime
            except OSError:
                continue
            if name in self.packages and self._mtimes.get(name) == mtime:
                continue
            info = resolve_package(init)
            with self._lock:
                pkg = self.packages.get(name)
                if pkg is None:
                    pkg = ManagedPackage(name=name)
                    self.packages[name] = pkg
                pkg.script_path = info.script_path
                pkg.target_version = info.target_version
                pkg.dependency = info.dependency
                pkg.total_time_seconds = info.total_time_seconds
            self._mtimes[name] = mtime
        # packages whose directory disappeared stop being reported
        gone = set(self.packages) - set(found)
        if gone:
            with self._lock:
                for name in gone:
                    self.packages.pop(name, None)
                    self._mtimes.pop(name, None)

    # -- runners ------------------------------------------------------------

    def _deps_ready(self, pkg: ManagedPackage) -> bool:
        for dep in pkg.dependency:
            name, minver = dep[0], dep[1]
            other = self.packages.get(name)
            if other is None or not other.is_installed:
                return False
            if minver != "*" and (not other.current_version
                                  or other.current_version < minver):
                return False
        return True

    def _do_install(self, pkg: ManagedPackage) -> None:
        res = _run_pkg(pkg.script_path, "install", stream=True)
        if res.exit_code == 0:
            start = _run_pkg(pkg.script_path, "start")
            if start.exit_code != 0:
                logger.error("package %s failed to start after install (%d)",
                             pkg.name, start.exit_code)
            with self._lock:
                pkg.is_installed = True
        else:
            logger.error("package %s install failed (%d)", pkg.name,
                         res.exit_code)
        with self._lock:
            pkg.installing = False
            pkg.progress = 100
            pkg.install_started = None

    def _install_pass(self) -> None:
        for pkg in list(self.packages.values()):
            if pkg.installing or pkg.is_installed:
                continue
            if not self._deps_ready(pkg):
                continue
            if _run_pkg(pkg.script_path, "shouldSkip", timeout=60).exit_code == 0:
                with self._lock:
                    pkg.skipped = True
                    pkg.is_installed = True
                    pkg.progress = 100
                continue
            if _run_pkg(pkg.script_path, "isInstalled", timeout=60).exit_code == 0:
                with self._lock:
                    pkg.is_installed = True
                    pkg.progress = 100
                continue
            with self._lock:
                pkg.installing = True
                pkg.progress = 0
                pkg.install_started = time.time()
            if self.inline_installs:
                self._do_install(pkg)
            else:
                threading.Thread(
                    target=self._do_install, args=(pkg,), daemon=True,
                    name=f"gpud-pkg-install-{pkg.name}",
                ).start()

    def _do_upgrade(self, pkg: ManagedPackage) -> None:
        res = _run_pkg(pkg.script_path, "upgrade", stream=True)
        if res.exit_code != 0:
            logger.error("package %s upgrade failed (%d)", pkg.name,
                         res.exit_code)
        with self._lock:
            pkg.installing = False
            pkg.progress = 100
            pkg.install_started = None

    def _update_pass(self) -> None:
        for pkg in list(self.packages.values()):
            if not pkg.is_installed or pkg.installing:
                continue
            ver = _run_pkg(pkg.script_path, "version", timeout=60)
            if ver.exit_code != 0 or not ver.output.strip():
                continue
            with self._lock:
                pkg.current_version = ver.output.strip().splitlines()[-1]
            if _run_pkg(pkg.script_path, "shouldSkip", timeout=60).exit_code == 0:
                with self._lock:
                    pkg.skipped = True
                continue
            if not pkg.target_version or pkg.current_version == pkg.target_version:
                continue
            logger.info("package %s version %s -> target %s: upgrading",
                        pkg.name, pkg.current_version, pkg.target_version)
            with self._lock:
                pkg.installing = True
                pkg.progress = 0
                pkg.install_started = time.time()
            if self.inline_installs:
                self._do_upgrade(pkg)
            else:
                threading.Thread(
                    target=self._do_upgrade, args=(pkg,), daemon=True,
                    name=f"gpud-pkg-upgrade-{pkg.name}",
                ).start()

    def _status_pass(self) -> None:
        for pkg in list(self.packages.values()):
            if not pkg.is_installed or pkg.installing:
                continue
            if _run_pkg(pkg.script_path, "shouldSkip", timeout=60).exit_code == 0:
                with self._lock:
                    pkg.skipped = True
                    pkg.status = True
                continue
            if _run_pkg(pkg.script_path, "status", timeout=60).exit_code == 0:
                with self._lock:
                    pkg.status = True
                continue
            with self._lock:
                pkg.status = False
            logger.warning("package %s status not ok: restarting", pkg.name)
            if _run_pkg(pkg.script_path, "stop").exit_code != 0:
                logger.error("package %s stop failed", pkg.name)
                continue
            if _run_pkg(pkg.script_path, "start").exit_code != 0:
                logger.error("package %s start failed", pkg.name)

    def _delete_pass(self) -> None:
        for pkg in list(self.packages.values()):
            if _run_pkg(pkg.script_path, "needDelete", timeout=60).exit_code != 0:
                continue
            if _run_pkg(pkg.script_path, "delete").exit_code != 0:
                logger.warning("package %s delete failed", pkg.name)

    def reconcile_once(self) -> None:
        """One full pass of every runner (install/update/status/delete).
        Used by the loop and, inline, by tests."""
        self._informer_pass()
        self._install_pass()
        self._update_pass()
        self._status_pass()
        self._delete_pass()
