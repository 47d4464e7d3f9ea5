"""Typed REST client for the gpud v1 API.

Reference: client/v1/v1.go:23-543 (GetHealthStates/GetEvents/GetMetrics/
GetInfo/...) and client/v1/healthz.go:62 (blocking healthz wait). The
daemon serves a self-signed certificate, so verification is off by default
like the reference's local client.
"""

from __future__ import annotations

import datetime
import time
from typing import Any, Dict, List, Optional

import httpx

from ..apiv1.types import Event, HealthState, Metric, rfc3339


class Client:
    def __init__(
        self,
        base_url: str = "https://localhost:15132",
        verify: bool = False,
        timeout: float = 15.0,
    ):
        self.base_url = base_url.rstrip("/")
        self._http = httpx.Client(verify=verify, timeout=timeout)

    def close(self) -> None:
        self._http.close()

    # -- raw helpers ---------------------------------------------------------

    def _get(self, path: str, params: Optional[Dict[str, str]] = None) -> Any:
        r = self._http.get(self.base_url + path, params=params)
        r.raise_for_status()
        return r.json()

    def _params(
        self,
        components: Optional[List[str]] = None,
        since: Optional[datetime.datetime] = None,
    ) -> Dict[str, str]:
        p: Dict[str, str] = {}
        if components:
            p["components"] = ",".join(components)
        if since is not None:
            p["startTime"] = rfc3339(since)
        return p

    # -- API -----------------------------------------------------------------

    def healthz(self) -> bool:
        try:
            r = self._http.get(self.base_url + "/healthz")
            return r.status_code == 200
        except httpx.HTTPError:
            return False

    def wait_healthz(self, timeout: float = 30.0) -> bool:
        deadline = time.time() + timeout
        while time.time() < deadline:
            if self.healthz():
                return True
            time.sleep(0.2)
        return False

    def get_components(self) -> List[str]:
        return self._get("/v1/components")

    def deregister_component(self, name: str) -> Any:
        r = self._http.delete(
            self.base_url + "/v1/components", params={"componentName": name}
        )
        r.raise_for_status()
        return r.json()

    def trigger_check(self, component: str = "", tag: str = "") -> Any:
        params = {}
        if component:
            params["componentName"] = component
        if tag:
            params["tagName"] = tag
        return self._get("/v1/components/trigger-check", params)

    def get_health_states(
        self, components: Optional[List[str]] = None
    ) -> Dict[str, List[HealthState]]:
        raw = self._get("/v1/states", self._params(components))
        return {
            item["component"]: [
                HealthState.from_dict(s) for s in item.get("states", [])
            ]
            for item in raw
        }

    def get_events(
        self,
        components: Optional[List[str]] = None,
        since: Optional[datetime.datetime] = None,
    ) -> Dict[str, List[Event]]:
        raw = self._get("/v1/events", self._params(components, since))
        return {
            item["component"]: [
                Event.from_dict(e) for e in item.get("events", [])
            ]
            for item in raw
        }

    def get_metrics(
        self,
        components: Optional[List[str]] = None,
        since: Optional[datetime.datetime] = None,
    ) -> Dict[str, List[Metric]]:
        raw = self._get("/v1/metrics", self._params(components, since))
        return {
            item["component"]: [
                Metric.from_dict(m) for m in item.get("metrics", [])
            ]
            for item in raw
        }

    def get_info(
        self,
        components: Optional[List[str]] = None,
        since: Optional[datetime.datetime] = None,
    ) -> Any:
        return self._get("/v1/info", self._params(components, since))

    def get_machine_info(self) -> Any:
        return self._get("/machine-info")

    def get_plugins(self) -> Any:
        return self._get("/v1/plugins")

    def set_healthy(self, components: Optional[List[str]] = None) -> Any:
        r = self._http.post(
            self.base_url + "/v1/health-states/set-healthy",
            params=self._params(components),
        )
        r.raise_for_status()
        return r.json()

    def inject_fault(
        self,
        ras_event_name: str = "",
        kernel_message: str = "",
        priority: int = 2,
    ) -> Any:
        body: Dict[str, Any] = {}
        if ras_event_name:
            body["ras_event_name"] = ras_event_name
        if kernel_message:
            body["kernel_message"] = {
                "message": kernel_message,
                "priority": priority,
            }
        r = self._http.post(self.base_url + "/inject-fault", json=body)
        r.raise_for_status()
        return r.json()

    def prometheus_metrics(self) -> str:
        r = self._http.get(self.base_url + "/metrics")
        r.raise_for_status()
        return r.text
