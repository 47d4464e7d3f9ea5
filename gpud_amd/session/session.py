"""Control-plane session (reference: pkg/session).

Two long-lived HTTP streams to the control plane (reference:
session.go:531 writer / :625 reader — JSON ``Body{Data, ReqID}`` frames),
a keepalive ticker (session_keepalive.go:11), jittered exponential
reconnect (session_reconnect.go:190), and a serve loop dispatching ~25
methods (session_process_request.go:24) against the daemon core.

The transport is pluggable (``open_reader`` / ``send_response`` injection)
so the dispatch and reconnect machinery is testable without a live control
plane — the reference tests do the same with httptest servers.
"""

from __future__ import annotations

import base64
import datetime
import json
import os
import random
import threading
import time
from typing import Callable, Dict, Iterator, List, Optional

import httpx

from ..apiv1.types import parse_rfc3339, utcnow
from ..bootstrap import DaemonCore
from ..pkg import custom_plugins, metadata
from ..pkg.fault_injector import Request as InjectRequest
from ..pkg.host import reboot_machine
from ..pkg.log import logger
from ..pkg.process_runner import run_bash

SESSION_PATH = "/api/v1/session"
KEEPALIVE_INTERVAL = 10.0
RECONNECT_BASE = 1.0
RECONNECT_MAX = 60.0


class Session:
    def __init__(
        self,
        core: DaemonCore,
        endpoint: str,
        token: str = "",
        machine_id: str = "",
        plugin_specs: Optional[List[custom_plugins.Spec]] = None,
        open_reader: Optional[Callable[[], Iterator[dict]]] = None,
        send_response: Optional[Callable[[dict], None]] = None,
        sleep_fn: Callable[[float], None] = time.sleep,
        jitter_fn: Callable[[], float] = lambda: random.uniform(0.5, 1.5),
        verify=None,
    ):
        self.core = core
        self.endpoint = endpoint.rstrip("/")
        self.token = token
        self.machine_id = machine_id
        self.plugin_specs = plugin_specs or []
        self._open_reader = open_reader or self._http_reader
        self._send_response = send_response or self._http_send
        self._sleep = sleep_fn
        self._jitter = jitter_fn
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        # verified TLS by default: this channel carries the auth token and
        # accepts bootstrap/setPluginSpecs/update — a MITM must not be able
        # to impersonate the control plane. Insecure mode exists only via
        # the explicit config flag (Config.control_plane_verify()).
        if verify is None:
            cfg = getattr(core, "config", None)
            verify = cfg.control_plane_verify() if cfg is not None else True
        self._client = httpx.Client(verify=verify, timeout=None)
        self.reconnects = 0

    # -- transport -----------------------------------------------------------

    def _headers(self) -> Dict[str, str]:
        return {
            "machine_id": self.machine_id,
            "token": self.token,
            "session_type": "reader",
        }

    def _http_reader(self) -> Iterator[dict]:
        """One reader-stream connection: yields request frames."""
        with self._client.stream(
            "POST",
            self.endpoint + SESSION_PATH,
            headers=self._headers(),
        ) as resp:
            resp.raise_for_status()
            for line in resp.iter_lines():
                if not line:
                    continue
                try:
                    yield json.loads(line)
                except json.JSONDecodeError:
                    continue

    def _http_send(self, frame: dict) -> None:
        self._client.post(
            self.endpoint + SESSION_PATH,
            headers={**self._headers(), "session_type": "writer"},
            json=frame,
        )

    # -- lifecycle -------------------------------------------------------------

    def start(self) -> None:
        self._thread = threading.Thread(
            target=self._serve_loop, daemon=True, name="gpud-session"
        )
        self._thread.start()
        # keepalive ticker (reference: session_keepalive.go:11) — a periodic
        # ping frame on the writer stream so the control plane sees liveness
        self._ka_thread = threading.Thread(
            target=self._keepalive_loop, daemon=True, name="gpud-session-ka"
        )
        self._ka_thread.start()

    def _keepalive_loop(self) -> None:
        while not self._stop.wait(KEEPALIVE_INTERVAL):
            try:
                self._send_response(
                    {"req_id": "", "method": "ping", "data": {"pong": True}}
                )
            except Exception:
                pass  # reconnect machinery handles the stream

    def stop(self) -> None:
        self._stop.set()
        self._client.close()

    def _serve_loop(self) -> None:
        backoff = RECONNECT_BASE
        while not self._stop.is_set():
            try:
                for frame in self._open_reader():
                    if self._stop.is_set():
                        return
                    backoff = RECONNECT_BASE  # healthy stream resets backoff
                    resp = self.process_request(frame)
                    if resp is not None:
                        try:
                            self._send_response(resp)
                        except Exception:
                            logger.exception("session response send failed")
            except Exception as e:
                if self._stop.is_set():
                    return
                logger.warning("session stream error: %s", e)
            self.reconnects += 1
            self._sleep(min(backoff * self._jitter(), RECONNECT_MAX))
            backoff = min(backoff * 2, RECONNECT_MAX)

    # -- dispatch (reference: session_process_request.go:24) -------------------

    def process_request(self, frame: dict) -> Optional[dict]:
        req_id = frame.get("req_id", "")
        method = frame.get("method", "")
        payload = frame.get("data") or {}
        if isinstance(payload, str):
            try:
                payload = json.loads(payload)
            except json.JSONDecodeError:
                payload = {}
        handler = getattr(self, f"_m_{method.replace('-', '_')}", None)
        if handler is None:
            body = {"error": f"unknown method {method!r}"}
        else:
            try:
                body = handler(payload)
            except Exception as e:
                logger.exception("session method %s failed", method)
                body = {"error": str(e)}
        return {"req_id": req_id, "method": method, "data": body}

    # -- methods ---------------------------------------------------------------

    def _since(self, payload: dict) -> datetime.datetime:
        raw = payload.get("since") or payload.get("startTime")
        if raw:
            t = parse_rfc3339(raw)
            if t:
                return t
        return utcnow() - datetime.timedelta(days=3)

    def _m_ping(self, payload: dict) -> dict:
        return {"pong": True}

    def _m_gossip(self, payload: dict) -> dict:
        from ..pkg.machine_info import get_machine_info

        return {"machineInfo": get_machine_info(self.core.smi_instance).to_dict()}

    def _m_states(self, payload: dict) -> dict:
        wanted = payload.get("components")
        out = []
        for comp in self.core.registry.all_components():
            if wanted and comp.name not in wanted:
                continue
            out.append(
                {
                    "component": comp.name,
                    "states": [s.to_dict() for s in comp.last_health_states()],
                }
            )
        return {"states": out}

    def _m_events(self, payload: dict) -> dict:
        since = self._since(payload)
        out = []
        for comp in self.core.registry.all_components():
            try:
                evs = comp.events(since)
            except Exception:
                evs = []
            if evs:
                out.append(
                    {
                        "component": comp.name,
                        "events": [e.to_dict() for e in evs],
                    }
                )
        return {"events": out}

    def _m_metrics(self, payload: dict) -> dict:
        by_comp = self.core.metrics_store.read(since=self._since(payload))
        return {
            "metrics": [
                {"component": c, "metrics": [m.to_dict() for m in ms]}
                for c, ms in sorted(by_comp.items())
            ]
        }

    def _m_triggerComponent(self, payload: dict) -> dict:
        return self._m_triggerComponentCheck(payload)

    def _m_triggerComponentCheck(self, payload: dict) -> dict:
        name = payload.get("component_name", "") or payload.get("componentName", "")
        tag = payload.get("tag_name", "") or payload.get("tagName", "")
        comps = []
        if name:
            comp = self.core.registry.get(name)
            if comp is None:
                return {"error": f"component {name!r} not found"}
            comps = [comp]
        elif tag:
            comps = [
                c for c in self.core.registry.all_components() if tag in c.tags()
            ]
        states = []
        for comp in comps:
            cr = comp.trigger_check()
            states.extend(s.to_dict() for s in cr.health_states())
        return {"states": states}

    def _m_deregisterComponent(self, payload: dict) -> dict:
        name = payload.get("component_name", "") or payload.get("componentName", "")
        comp = self.core.registry.get(name)
        if comp is None:
            return {"error": f"component {name!r} not found"}
        if not comp.deregisterable():
            return {"error": f"component {name!r} is not deregisterable"}
        comp.close()
        self.core.registry.deregister(name)
        return {"deregistered": name}

    def _m_setHealthy(self, payload: dict) -> dict:
        wanted = payload.get("components")
        done = []
        for comp in self.core.registry.all_components():
            if wanted and comp.name not in wanted:
                continue
            if comp.can_set_healthy():
                comp.set_healthy()
                done.append(comp.name)
        return {"set_healthy": done}

    def _m_injectFault(self, payload: dict) -> dict:
        err = self.core.fault_injector.inject(InjectRequest.from_dict(payload))
        return {"error": err} if err else {"status": "injected"}

    def _m_setPluginSpecs(self, payload: dict) -> dict:
        specs = [custom_plugins.Spec.from_dict(d) for d in payload.get("specs", [])]
        registered, errors = [], []
        for spec in specs:
            err = spec.validate()
            if err:
                errors.append(err)
                continue
            for comp in custom_plugins.make_components(spec):
                existing = self.core.registry.get(comp.name)
                if existing is not None:
                    existing.close()
                    self.core.registry.deregister(comp.name)
                self.core.registry.register_component(comp)
                comp.start()
                registered.append(comp.name)
        self.plugin_specs = specs
        return {"registered": registered, "errors": errors}

    def _m_getPluginSpecs(self, payload: dict) -> dict:
        return {"specs": [s.to_dict() for s in self.plugin_specs]}

    def _m_packageStatus(self, payload: dict) -> dict:
        try:
            from ..pkg.gpud_manager import package_statuses

            return {
                "packages": [
                    p.to_dict() for p in package_statuses(self.core.config)
                ]
            }
        except Exception as e:
            return {"packages": [], "error": str(e)}

    DIAG_COMPONENTS = (
        "accelerator-amd-diag-mfma",
        "accelerator-amd-diag-bandwidth",
        "accelerator-amd-diag-fabric",
    )

    def _m_diagnostic(self, payload: dict) -> dict:
        """Run the active diagnostics (the reference runs its diagnostic
        asynchronously and uploads — diagnostic.go:48; we run the CDNA4
        diag components). ``{"async": true}`` starts them in the
        background — results land in each component's cached health state
        (queryable via the ``states`` method or /v1/states)."""
        if payload.get("report_id"):
            # managed-diagnostic protocol (reference docs/INTEGRATION.md +
            # pkg/session/diagnostic.go): the request carries report_id, a
            # FIXED type and timeout_seconds — never user commands. We
            # accept the amd_bug_report analog, run the bundle collection
            # asynchronously under the timeout, upload to the presigned
            # URL, and notify the control plane's failure endpoint on any
            # failure (collection error, timeout, no artifact, upload
            # failure after it produced nothing).
            report_id = str(payload["report_id"])
            dtype = payload.get("type", "amd_bug_report")
            if dtype not in ("amd_bug_report", "nvidia_bug_report"):
                return {"error": f"unsupported diagnostic type {dtype!r}"}
            timeout_s = float(payload.get("timeout_seconds", 600))
            upload_url = payload.get("upload_url", "")

            def _notify_failure() -> None:
                if not self.endpoint:
                    return
                try:
                    self._client.post(
                        f"{self.endpoint}/api/v1/diagnostics/{report_id}/failure",
                        headers=self._headers(),
                        json={"report_id": report_id},
                        timeout=30,
                    )
                except Exception:
                    logger.exception("diagnostic failure notification failed")

            def _run_report():
                import tempfile

                from ..pkg.bundle import collect_bundle, upload_bundle

                path = os.path.join(
                    tempfile.gettempdir(), f"gpud-diag-{report_id}.tar.gz"
                )
                done = threading.Event()
                err_box: list = []

                def _collect():
                    try:
                        collect_bundle(path, core=self.core)
                    except Exception as e:  # noqa: BLE001
                        err_box.append(str(e))
                    finally:
                        done.set()

                t = threading.Thread(target=_collect, daemon=True)
                t.start()
                if not done.wait(timeout_s) or err_box or not os.path.exists(path):
                    logger.warning("diagnostic %s failed/timed out", report_id)
                    _notify_failure()
                    return
                if upload_url:
                    err = upload_bundle(path, upload_url)
                    if err:
                        logger.warning("diagnostic %s upload: %s", report_id, err)
                        _notify_failure()

            threading.Thread(
                target=_run_report, daemon=True, name=f"gpud-diag-{report_id}"
            ).start()
            return {"status": "accepted", "report_id": report_id, "type": dtype}
        if payload.get("bundle"):
            # direct support-bundle mode (synchronous; operator-driven)
            import tempfile

            from ..pkg.bundle import collect_bundle, upload_bundle

            path = payload.get("path") or os.path.join(
                tempfile.gettempdir(), "gpud-bundle.tar.gz"
            )
            collect_bundle(path, core=self.core)
            out = {"bundle": path, "size": os.path.getsize(path)}
            url = payload.get("upload_url", "")
            if url:
                err = upload_bundle(path, url)
                out["uploaded"] = err is None
                if err:
                    out["upload_error"] = err
            return out
        if payload.get("async"):
            def _run():
                for name in self.DIAG_COMPONENTS:
                    comp = self.core.registry.get(name)
                    if comp is not None:
                        comp.trigger_check()

            threading.Thread(
                target=_run, daemon=True, name="gpud-diagnostic"
            ).start()
            return {"status": "started", "components": list(self.DIAG_COMPONENTS)}
        results = {}
        for name in self.DIAG_COMPONENTS:
            comp = self.core.registry.get(name)
            if comp is None:
                continue
            cr = comp.trigger_check()
            results[name] = {
                "health": cr.health,
                "reason": cr.reason,
                "extra": cr.extra_info or {},
            }
        return {"diagnostics": results}

    def _m_bootstrap(self, payload: dict) -> dict:
        script_b64 = payload.get("script", "")
        try:
            script = base64.b64decode(script_b64).decode()
        except Exception:
            return {"error": "script must be base64"}
        timeout = float(payload.get("timeout_seconds", 120))
        res = run_bash(script, timeout_seconds=timeout)
        return {"exit_code": res.exit_code, "output": res.output[-4096:]}

    # Config keys the control plane may update at runtime (reference
    # restricts updateConfig to specific known keys —
    # pkg/session/session.go:223-233). Command overrides (reboot_command
    # etc.), data paths and TLS settings are deliberately NOT settable over
    # the session: those would turn a config push into code execution or a
    # downgrade of the channel's own security.
    UPDATABLE_CONFIG_KEYS = frozenset(
        {
            "events_retention_days",
            "metrics_retention_days",
            "compact_period_hours",
            "poll_interval_seconds",
            "metrics_sync_interval_seconds",
            "enabled_components",
            "disabled_components",
            "expected_gpu_count",
            "expected_xgmi_link_count",
            "expected_compute_partition",
            "expected_memory_partition",
            "expected_ib_ports",
            "expected_ib_rate_gbps",
            "latency_targets",
            "temperature_margin_threshold_c",
            "ras_reboot_threshold",
            "ras_event_thresholds",
            "zombie_degraded_threshold",
            "zombie_unhealthy_threshold",
            "dstate_persistence_threshold",
            "dstate_name_regexes",
            "nfs_host_root",
            "kernel_modules_to_check",
            "libraries_to_check",
            "mount_points",
        }
    )

    def _m_updateConfig(self, payload: dict) -> dict:
        applied, rejected = [], []
        cfg = self.core.config
        for key, value in payload.items():
            if key in self.UPDATABLE_CONFIG_KEYS and hasattr(cfg, key):
                setattr(cfg, key, value)
                applied.append(key)
            else:
                rejected.append(key)
        # poll interval reaches running tickers immediately — the loop
        # re-reads component.poll_interval every cycle (reference:
        # updateConfig pushes SetDefault* setters live)
        registry = getattr(self.core, "registry", None)
        if "poll_interval_seconds" in applied and registry is not None:
            try:
                iv = float(cfg.poll_interval_seconds)
                if iv > 0:
                    for comp in registry.all_components():
                        if hasattr(comp, "poll_interval"):
                            comp.poll_interval = iv
            except (TypeError, ValueError):
                pass
        out = {"applied": applied}
        if rejected:
            out["rejected"] = rejected
        return out

    def _m_updateToken(self, payload: dict) -> dict:
        token = payload.get("token", "")
        if not token:
            return {"error": "token required"}
        self.token = token
        if self.core.db_rw is not None:
            metadata.set_value(self.core.db_rw, metadata.KEY_TOKEN, token)
        return {"status": "token updated"}

    def _m_getToken(self, payload: dict) -> dict:
        return {"token": self.token}

    def _m_reboot(self, payload: dict) -> dict:
        delay = float(payload.get("delay_seconds", 0))

        def _do():
            if delay:
                time.sleep(delay)
            err = reboot_machine(self.core.config.reboot_command)
            if err:
                logger.error("reboot failed: %s", err)

        threading.Thread(target=_do, daemon=True).start()
        return {"status": "reboot scheduled"}

    def _m_update(self, payload: dict) -> dict:
        from ..pkg.update import update_to_version

        ver = payload.get("version", "")
        if not ver:
            return {"error": "version required"}
        err = update_to_version(self.core.config, ver)
        return {"error": err} if err else {"status": f"updating to {ver}"}

    def _m_logout(self, payload: dict) -> dict:
        if self.core.db_rw is not None:
            metadata.delete_value(self.core.db_rw, metadata.KEY_TOKEN)
            metadata.delete_value(self.core.db_rw, metadata.KEY_MACHINE_PROOF)
        self.token = ""
        return {"status": "logged out"}

    def _m_delete(self, payload: dict) -> dict:
        return self._m_logout(payload)

    # -- kap-mTLS (reference: pkg/kapmtls + session methods) ------------------

    def _kapmtls(self):
        from ..pkg.kapmtls import Manager

        import os

        return Manager(os.path.join(self.core.config.data_dir, "kapmtls"))

    def _m_kapMTLSStatus(self, payload: dict) -> dict:
        return self._kapmtls().status()

    def _m_updateKAPMTLSCredentials(self, payload: dict) -> dict:
        cert = base64.b64decode(payload.get("cert", ""))
        key = base64.b64decode(payload.get("key", ""))
        if not cert or not key:
            return {"error": "cert and key (base64) required"}
        version = self._kapmtls().stage(
            cert,
            key,
            payload.get("version", ""),
            gateway_ca_pem=base64.b64decode(payload.get("gateway_ca", "")),
            gateway_endpoint=payload.get("gateway_endpoint", ""),
            server_name=payload.get("server_name", ""),
        )
        return {"staged_version": version}

    def _m_activateKAPMTLS(self, payload: dict) -> dict:
        mgr = self._kapmtls()
        err = mgr.activate(payload.get("version", ""))
        if err:
            return {"error": err}
        return {"active_version": mgr.active_version()}

    # nodeCredentials: the control plane places identity files on the node
    # (reference: pkg/session/node_credentials.go). The destination travels
    # with the file, so the allow-list below is the entire safety boundary:
    # without it this method is an arbitrary root file write. These trees
    # hold node identity material and nothing that grants code execution.
    NODE_CREDENTIAL_ALLOWED_PREFIXES = ["/var/lib/gpud/", "/etc/kubernetes/"]
    DEFAULT_NODE_CREDENTIAL_MODE = 0o600

    def _allowed_credential_prefixes(self) -> List[str]:
        return list(self.NODE_CREDENTIAL_ALLOWED_PREFIXES)

    def _validate_credential_file(self, f: dict) -> Optional[str]:
        path = f.get("path", "")
        if not path:
            return "node credential path is required"
        if not os.path.isabs(path):
            return f"node credential path {path!r} must be absolute"
        # compared after normalization so '..' cannot escape an allowed tree
        cleaned = os.path.normpath(path)
        prefixes = self._allowed_credential_prefixes()
        if not any(cleaned.startswith(p) for p in prefixes):
            return (
                f"node credential path {path!r} is outside "
                + ", ".join(prefixes)
            )
        if not f.get("contents"):
            return f"node credential {path!r} has no contents"
        return None

    def _write_credential_file(self, f: dict) -> Optional[str]:
        """Atomic publish: staged in the destination directory with the
        final mode, then renamed — a reader sees old or new contents, never
        a partial write, and the material is never briefly world-readable."""
        path = os.path.normpath(f["path"])
        contents = base64.b64decode(f["contents"])
        mode = int(f.get("mode") or 0) or self.DEFAULT_NODE_CREDENTIAL_MODE
        try:
            os.makedirs(os.path.dirname(path), exist_ok=True)
            staged = path + ".gpud-staged"
            fd = os.open(staged, os.O_WRONLY | os.O_CREAT | os.O_TRUNC, mode)
            try:
                os.write(fd, contents)
            finally:
                os.close(fd)
            os.chmod(staged, mode)
            os.replace(staged, path)
        except OSError as e:
            return f"writing node credential {path!r}: {e}"
        return None

    def _m_nodeCredentials(self, payload: dict) -> dict:
        kubelet = payload.get("kubelet") or {}
        files = [
            f
            for f in (kubelet.get("config"), kubelet.get("client_certificate"))
            if f
        ]
        if not files:
            return {"error": "node credentials contain no files"}
        # validate everything before writing anything: one bad path leaves
        # the node exactly as it was, not half-updated
        for f in files:
            err = self._validate_credential_file(f)
            if err:
                return {"error": err}
        written = []
        for f in files:
            err = self._write_credential_file(f)
            if err:
                return {"error": err, "written": written}
            written.append(os.path.normpath(f["path"]))
        return {"written": written}
