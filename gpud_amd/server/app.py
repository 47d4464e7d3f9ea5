"""HTTP API (reference: pkg/server/handlers_components.go, server.go routes).

Routes kept URL- and wire-compatible with the reference (SURVEY.md
appendix A HTTP surface): ``/v1/components`` GET+DELETE,
``/v1/components/trigger-check``, ``/v1/components/trigger-tag``,
``/v1/states``, ``/v1/events``, ``/v1/info``, ``/v1/metrics``,
``/v1/health-states/set-healthy`` POST, ``/v1/plugins``, ``/healthz``,
``/machine-info``, ``/inject-fault`` POST, ``/metrics`` (Prometheus),
``/admin/config``. Content negotiation: JSON default, YAML via
``Content-Type: application/yaml`` request header, indented JSON via
``json-indent: true`` (reference: handlers_components.go:300-318).
"""

from __future__ import annotations

import datetime
import json
from typing import Any, List, Optional

import yaml
from fastapi import FastAPI, Request, Response
from prometheus_client import generate_latest, CONTENT_TYPE_LATEST

from .. import __version__
from ..apiv1.types import parse_rfc3339, utcnow
from ..bootstrap import DaemonCore
from ..pkg.fault_injector import Request as InjectRequest
from ..pkg.log import logger


def _negotiate(request: Request, payload: Any, status_code: int = 200) -> Response:
    ct = request.headers.get("content-type", "")
    if ct == "application/yaml":
        return Response(
            content=yaml.safe_dump(payload, sort_keys=False),
            media_type="application/yaml",
            status_code=status_code,
        )
    indent = 2 if request.headers.get("json-indent") == "true" else None
    return Response(
        content=json.dumps(payload, indent=indent),
        media_type="application/json",
        status_code=status_code,
    )


def _err(request: Request, code: int, message: str) -> Response:
    return _negotiate(request, {"code": code, "message": message}, code)


def _parse_components_param(raw: Optional[str]) -> Optional[List[str]]:
    if not raw:
        return None
    return [c for c in raw.split(",") if c]


def _parse_since(request: Request) -> datetime.datetime:
    raw = request.query_params.get("startTime")
    if raw:
        try:
            return parse_rfc3339(raw)
        except ValueError:
            pass
    return utcnow() - datetime.timedelta(days=3)


def create_app(core: DaemonCore, plugin_specs: Optional[list] = None) -> FastAPI:
    app = FastAPI(title="gpud", version=__version__, docs_url="/swagger")
    # gzip via Accept-Encoding (reference: pkg/server/server.go:412)
    from starlette.middleware.gzip import GZipMiddleware

    app.add_middleware(GZipMiddleware, minimum_size=1024)
    registry = core.registry
    app.state.core = core
    app.state.plugin_specs = plugin_specs or []

    # -- health -------------------------------------------------------------

    @app.get("/healthz")
    def healthz():
        return {"status": "ok", "version": "v1"}

    # -- /v1/components ------------------------------------------------------

    @app.get("/v1/components")
    def get_components(request: Request):
        return _negotiate(request, sorted(registry.names()))

    @app.delete("/v1/components")
    def deregister_component(request: Request):
        name = request.query_params.get("componentName", "")
        if not name:
            return _err(request, 400, "componentName is required")
        comp = registry.get(name)
        if comp is None:
            return _err(request, 404, f"component {name!r} not found")
        if not comp.deregisterable():
            return _err(request, 400, f"component {name!r} is not deregisterable")
        comp.close()
        registry.deregister(name)
        return _negotiate(request, {"message": "deregistered", "component": name})

    @app.get("/v1/components/trigger-check")
    def trigger_check(request: Request):
        name = request.query_params.get("componentName", "")
        tag = request.query_params.get("tagName", "")
        if bool(name) == bool(tag):
            return _err(
                request, 400, "exactly one of componentName or tagName required"
            )
        comps = []
        if name:
            comp = registry.get(name)
            if comp is None:
                return _err(request, 404, f"component {name!r} not found")
            comps = [comp]
        else:
            comps = [c for c in registry.all_components() if tag in c.tags()]
        states = []
        for comp in comps:
            cr = comp.trigger_check()
            states.extend(s.to_dict() for s in cr.health_states())
        return _negotiate(request, {"states": states})

    @app.get("/v1/components/trigger-tag")
    def trigger_tag(request: Request):
        tag = request.query_params.get("tagName", "")
        if not tag:
            return _err(request, 400, "tagName is required")
        triggered, ok = [], True
        for comp in registry.all_components():
            if tag in comp.tags():
                cr = comp.trigger_check()
                triggered.append(comp.name)
                if cr.health_state_type() != "Healthy":
                    ok = False
        return _negotiate(
            request,
            {"components": triggered, "exit": 0 if ok else 1, "success": ok},
        )

    # -- states / events / info / metrics ------------------------------------

    @app.get("/v1/states")
    def get_states(request: Request):
        wanted = _parse_components_param(request.query_params.get("components"))
        out = []
        for comp in registry.all_components():
            if wanted and comp.name not in wanted:
                continue
            out.append(
                {
                    "component": comp.name,
                    "states": [s.to_dict() for s in comp.last_health_states()],
                }
            )
        return _negotiate(request, out)

    @app.get("/v1/events")
    def get_events(request: Request):
        wanted = _parse_components_param(request.query_params.get("components"))
        since = _parse_since(request)
        now = utcnow()
        out = []
        for comp in registry.all_components():
            if wanted and comp.name not in wanted:
                continue
            try:
                evs = comp.events(since)
            except Exception as e:
                logger.warning("events(%s) failed: %s", comp.name, e)
                evs = []
            out.append(
                {
                    "component": comp.name,
                    "startTime": since.strftime("%Y-%m-%dT%H:%M:%SZ"),
                    "endTime": now.strftime("%Y-%m-%dT%H:%M:%SZ"),
                    "events": [e.to_dict() for e in evs],
                }
            )
        return _negotiate(request, out)

    @app.get("/v1/metrics")
    def get_metrics(request: Request):
        wanted = _parse_components_param(request.query_params.get("components"))
        since = _parse_since(request)
        by_comp = core.metrics_store.read(since=since, components=wanted)
        out = [
            {"component": comp, "metrics": [m.to_dict() for m in ms]}
            for comp, ms in sorted(by_comp.items())
        ]
        return _negotiate(request, out)

    @app.get("/v1/info")
    def get_info(request: Request):
        wanted = _parse_components_param(request.query_params.get("components"))
        since = _parse_since(request)
        now = utcnow()
        metrics_by_comp = core.metrics_store.read(since=since, components=wanted)
        out = []
        for comp in registry.all_components():
            if wanted and comp.name not in wanted:
                continue
            try:
                evs = comp.events(since)
            except Exception:
                evs = []
            out.append(
                {
                    "component": comp.name,
                    "startTime": since.strftime("%Y-%m-%dT%H:%M:%SZ"),
                    "endTime": now.strftime("%Y-%m-%dT%H:%M:%SZ"),
                    "info": {
                        "states": [
                            s.to_dict() for s in comp.last_health_states()
                        ],
                        "events": [e.to_dict() for e in evs],
                        "metrics": [
                            m.to_dict()
                            for m in metrics_by_comp.get(comp.name, [])
                        ],
                    },
                }
            )
        return _negotiate(request, out)

    @app.post("/v1/health-states/set-healthy")
    def set_healthy(request: Request):
        wanted = _parse_components_param(request.query_params.get("components"))
        done, skipped = [], []
        for comp in registry.all_components():
            if wanted and comp.name not in wanted:
                continue
            if comp.can_set_healthy():
                comp.set_healthy()
                done.append(comp.name)
            elif wanted:
                skipped.append(comp.name)
        return _negotiate(request, {"set_healthy": done, "skipped": skipped})

    @app.get("/v1/plugins")
    def get_plugins(request: Request):
        return _negotiate(
            request,
            [
                s.to_dict() if hasattr(s, "to_dict") else s
                for s in app.state.plugin_specs
            ],
        )

    # -- machine info / fault injection / admin -------------------------------

    @app.get("/machine-info")
    def machine_info(request: Request):
        from ..pkg.machine_info import get_machine_info

        return _negotiate(request, get_machine_info(core.smi_instance).to_dict())

    @app.post("/inject-fault")
    async def inject_fault(request: Request):
        try:
            body = await request.json()
        except Exception:
            return _err(request, 400, "invalid JSON body")
        req = InjectRequest.from_dict(body or {})
        err = core.fault_injector.inject(req)
        if err:
            return _err(request, 400, err)
        return _negotiate(request, {"status": "injected"})

    @app.get("/admin/config")
    def admin_config(request: Request):
        return _negotiate(request, core.config.to_dict())

    @app.get("/admin/packages")
    def admin_packages(request: Request):
        # prefer the live controller's rich status (reference: the manager's
        # packages.PackageStatus shape); fall back to one-shot probes
        ctl = getattr(core, "pkg_controller", None)
        if ctl is not None:
            return _negotiate(request, ctl.admin_statuses())
        from ..pkg.gpud_manager import package_statuses

        return _negotiate(
            request, [p.to_dict() for p in package_statuses(core.config)]
        )

    # -- profiling (reference: /admin/pprof/* behind --pprof;
    #    Python analog: live thread stacks + tracemalloc heap) --------------

    @app.get("/admin/pprof/threads")
    def pprof_threads(request: Request):
        import sys
        import traceback

        frames = sys._current_frames()
        out = {}
        import threading as _t

        names = {t.ident: t.name for t in _t.enumerate()}
        for tid, frame in frames.items():
            out[names.get(tid, str(tid))] = traceback.format_stack(frame)
        return _negotiate(request, out)

    @app.get("/admin/pprof/profile")
    def pprof_profile(request: Request, seconds: float = 5.0):
        """CPU profile analog of the reference's /admin/pprof/profile
        (pkg/server/server.go:435-440): a statistical sampler over
        sys._current_frames() — samples every thread's stack at 10 ms for
        the requested window and reports functions by inclusive sample
        count (py-spy-style, no instrumentation overhead between samples)."""
        import collections
        import sys
        import time as _time

        seconds = max(0.1, min(seconds, 30.0))
        interval = 0.01
        own = {__import__("threading").get_ident()}
        counts: "collections.Counter[str]" = collections.Counter()
        leaf_counts: "collections.Counter[str]" = collections.Counter()
        n_samples = 0
        deadline = _time.monotonic() + seconds
        while _time.monotonic() < deadline:
            for tid, frame in sys._current_frames().items():
                if tid in own:
                    continue
                f = frame
                leaf = True
                while f is not None:
                    key = (
                        f"{f.f_code.co_filename.rsplit('/', 1)[-1]}:"
                        f"{f.f_code.co_name}"
                    )
                    counts[key] += 1
                    if leaf:
                        leaf_counts[key] += 1
                        leaf = False
                    f = f.f_back
            n_samples += 1
            _time.sleep(interval)
        return _negotiate(
            request,
            {
                "seconds": seconds,
                "samples": n_samples,
                "top_inclusive": [
                    {"func": k, "samples": v}
                    for k, v in counts.most_common(50)
                ],
                "top_self": [
                    {"func": k, "samples": v}
                    for k, v in leaf_counts.most_common(50)
                ],
            },
        )

    @app.get("/admin/pprof/heap")
    def pprof_heap(request: Request):
        import tracemalloc

        if not tracemalloc.is_tracing():
            tracemalloc.start()
            return _negotiate(
                request, {"status": "tracing started; query again for stats"}
            )
        snap = tracemalloc.take_snapshot()
        top = snap.statistics("lineno")[:50]
        return _negotiate(request, [str(s) for s in top])

    # -- prometheus -----------------------------------------------------------

    @app.get("/metrics")
    def prometheus_metrics():
        return Response(
            content=generate_latest(core.metrics_registry),
            media_type=CONTENT_TYPE_LATEST,
        )

    return app
