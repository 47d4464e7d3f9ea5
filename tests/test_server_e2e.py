"""E2E: boot the real daemon (mock SMI) and exercise the HTTP surface.

Reference: e2e/e2e_test.go:35-710 — real daemon on a random port with
mocked GPU layer + mocked lspci, exercised over HTTPS: healthz,
machine-info, /v1/states, set-healthy, plugins, /v1/metrics, /metrics.
"""

import datetime
import json
import socket
import time

import pytest

from gpud_amd.apiv1.types import HealthStateType, utcnow


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture(scope="module")
def daemon(tmp_path_factory):
    import os

    os.environ["GPUD_AMDSMI_MOCK"] = "1"
    os.environ["GPUD_AMDSMI_MOCK_GPUS"] = "2"
    from gpud_amd.bootstrap import build_core
    from gpud_amd.client import Client
    from gpud_amd.pkg.config import Config
    from gpud_amd.pkg.custom_plugins import Spec
    from gpud_amd.server import Server

    cfg = Config(data_dir=str(tmp_path_factory.mktemp("gpud-data")))
    core = build_core(cfg, in_memory_db=True, kmsg_writable=False)
    # prime all component states once (like scan); don't start tickers to
    # keep the test deterministic
    for c in core.registry.all_components():
        if getattr(c, "run_mode", "") != "manual":
            c.trigger_check()
    core.metrics_syncer.sync_once()
    specs = [
        Spec.from_dict(
            {
                "plugin_name": "exampleplugin",
                "plugin_type": "component",
                "health_state_plugin": {
                    "steps": [{"run_bash_script": {"script": "echo ok"}}]
                },
            }
        )
    ]
    port = _free_port()
    server = Server(core, port=port, tls=True, plugin_specs=specs)
    server.start()
    client = Client(server.base_url)
    assert client.wait_healthz(15)
    yield core, server, client
    client.close()
    server.stop()
    core.close()
    os.environ.pop("GPUD_AMDSMI_MOCK", None)


def test_healthz_and_components(daemon):
    core, server, client = daemon
    assert server.base_url.startswith("https://")  # self-signed TLS
    comps = client.get_components()
    assert "accelerator-amd-temperature" in comps
    assert "cpu" in comps and "os" in comps


def test_states_endpoint(daemon):
    _core, _server, client = daemon
    states = client.get_health_states()
    assert "accelerator-amd-temperature" in states
    st = states["accelerator-amd-temperature"][0]
    assert st.health == HealthStateType.HEALTHY
    # filter param works
    only = client.get_health_states(components=["cpu"])
    assert list(only.keys()) == ["cpu"]


def test_trigger_check_endpoint(daemon):
    _core, _server, client = daemon
    out = client.trigger_check(component="accelerator-amd-power")
    assert out["states"][0]["health"] == "Healthy"


def test_metrics_endpoints(daemon):
    _core, _server, client = daemon
    by_comp = client.get_metrics(since=utcnow() - datetime.timedelta(hours=1))
    assert "accelerator-amd-temperature" in by_comp
    prom = client.prometheus_metrics()
    assert "accelerator_amd_temperature_hotspot_celsius" in prom
    assert "gpud_component_check_duration_seconds" in prom


def test_events_and_info(daemon):
    _core, _server, client = daemon
    evs = client.get_events()
    assert "os" in evs  # reboot event bucket exists
    info = client.get_info(components=["cpu"])
    assert info[0]["component"] == "cpu"
    assert info[0]["info"]["states"]


def test_machine_info(daemon):
    _core, _server, client = daemon
    mi = client.get_machine_info()
    assert mi["hostname"]
    assert mi["gpuInfo"]["product"] == "AMD Instinct MI355X"
    assert len(mi["gpuInfo"]["gpus"]) == 2


def test_plugins_endpoint(daemon):
    _core, _server, client = daemon
    plugins = client.get_plugins()
    assert plugins[0]["plugin_name"] == "exampleplugin"


def test_inject_fault_roundtrip(daemon):
    core, _server, client = daemon
    out = client.inject_fault(ras_event_name="amdgpu_ring_timeout")
    assert out["status"] == "injected"
    # NoopWriter records it (kmsg not writable in tests)
    written = core.fault_injector.kmsg_writer().written
    assert any("timeout" in m for _p, m in written)


def test_set_healthy_endpoint(daemon):
    _core, _server, client = daemon
    out = client.set_healthy(["accelerator-amd-error-ras"])
    assert "accelerator-amd-error-ras" in out["set_healthy"]


def test_content_negotiation_yaml(daemon):
    _core, server, client = daemon
    import httpx

    r = httpx.get(
        server.base_url + "/v1/components",
        headers={"content-type": "application/yaml"},
        verify=False,
    )
    assert r.status_code == 200
    assert r.headers["content-type"].startswith("application/yaml")
    import yaml as y

    assert "cpu" in y.safe_load(r.text)


def test_deregister_rules(daemon):
    _core, server, client = daemon
    import httpx

    # built-ins are not deregisterable
    r = httpx.delete(
        server.base_url + "/v1/components",
        params={"componentName": "cpu"},
        verify=False,
    )
    assert r.status_code == 400
    r = httpx.delete(
        server.base_url + "/v1/components",
        params={"componentName": "nope"},
        verify=False,
    )
    assert r.status_code == 404


def test_trigger_tag_endpoint(daemon):
    _core, server, client = daemon
    import httpx

    r = httpx.get(
        server.base_url + "/v1/components/trigger-tag",
        params={"tagName": "network"},
        verify=False,
    )
    assert r.status_code == 200
    body = r.json()
    assert "components" in body and "success" in body


def test_gzip_negotiation(daemon):
    _core, server, _client = daemon
    import httpx

    r = httpx.get(
        server.base_url + "/v1/states",
        headers={"accept-encoding": "gzip"},
        verify=False,
    )
    assert r.status_code == 200
    # httpx transparently decompresses; verify the server really gzipped
    assert r.headers.get("content-encoding") == "gzip"


def test_admin_endpoints(daemon):
    _core, server, _client = daemon
    import httpx

    r = httpx.get(server.base_url + "/admin/config", verify=False)
    assert r.status_code == 200 and "data_dir" in r.json()
    r = httpx.get(server.base_url + "/admin/pprof/threads", verify=False)
    assert r.status_code == 200
    assert any("gpud" in k or "MainThread" in k for k in r.json())
    r = httpx.get(server.base_url + "/admin/packages", verify=False)
    assert r.status_code == 200
    # CPU-profile analog (reference: /admin/pprof/profile): sampled stacks
    r = httpx.get(
        server.base_url + "/admin/pprof/profile?seconds=0.3",
        verify=False, timeout=30,
    )
    assert r.status_code == 200
    prof = r.json()
    assert prof["samples"] > 0
    assert isinstance(prof["top_inclusive"], list)
    # swagger UI route + openapi schema (reference: /swagger/*any)
    r = httpx.get(server.base_url + "/swagger", verify=False)
    assert r.status_code == 200 and "swagger" in r.text.lower()
    r = httpx.get(server.base_url + "/openapi.json", verify=False)
    assert r.status_code == 200 and "/v1/states" in r.text


def test_malformed_requests_never_500(daemon):
    """Adversarial/garbage inputs degrade to 4xx, never 5xx (the control
    plane retries 5xx; a junk query param must not look like an outage)."""
    import httpx

    core, server, client = daemon
    base = server.base_url
    h = httpx.Client(verify=False, timeout=10)
    cases = [
        ("GET", "/v1/states", {"components": "no-such-component"}),
        ("GET", "/v1/states", {"components": "../../etc/passwd"}),
        ("GET", "/v1/events", {"startTime": "not-a-time"}),
        ("GET", "/v1/events", {"startTime": "99999999999999999999"}),
        ("GET", "/v1/metrics", {"since": "-1h%00"}),
        ("GET", "/v1/components/trigger-check", {"componentName": "\x00weird"}),
        ("GET", "/v1/components/trigger-tag", {"tagName": "'; DROP TABLE x;--"}),
        ("GET", "/v1/info", {"components": ","}),
    ]
    for method, path, params in cases:
        r = h.request(method, base + path, params=params)
        assert r.status_code < 500, (path, params, r.status_code, r.text[:200])
    # malformed bodies on POST endpoints
    for path, body in [
        ("/v1/health-states/set-healthy", b"{broken json"),
        ("/inject-fault", b"\xff\xfe binary"),
    ]:
        r = h.post(
            base + path, content=body, headers={"Content-Type": "application/json"}
        )
        assert r.status_code < 500, (path, r.status_code, r.text[:200])
    h.close()


def test_v1_metrics_includes_new_component_gauges(daemon):
    """The newest components' gauges survive the scrape→store→/v1/metrics
    path (the gpud_component label filter must not drop them)."""
    core, server, client = daemon
    core.registry.get("accelerator-amd-pcie").trigger_check()
    core.registry.get("accelerator-amd-partition").trigger_check()
    core.metrics_syncer.sync_once()
    metrics = client.get_metrics()
    names = {m.name for comp in metrics.values() for m in comp}
    assert any("pcie_link_width" in n for n in names), sorted(names)[:20]
    assert any("partition_count" in n for n in names)


def test_client_error_paths(daemon):
    """Client surfaces server 4xx as exceptions with useful messages, and
    wait_healthz times out cleanly on a dead port."""
    import httpx

    from gpud_amd.client import Client

    core, server, client = daemon
    with pytest.raises(httpx.HTTPStatusError):
        client.deregister_component("no-such-component")
    with pytest.raises(httpx.HTTPStatusError):
        client.trigger_check(component="also-missing")
    dead = Client("https://127.0.0.1:9")  # discard port, nothing listens
    assert dead.wait_healthz(timeout=1.5) is False
    dead.close()


def test_trigger_tag_unknown_tag_empty(daemon):
    core, server, client = daemon
    out = client.trigger_check(tag="no-such-tag")
    assert out.get("components", []) == [] or out.get("states", []) == []


def test_gzip_large_response(daemon):
    """Accept-Encoding: gzip compresses the full states payload (the
    GZip middleware the reference enables for /v1 routes)."""
    import httpx

    core, server, client = daemon
    h = httpx.Client(verify=False, timeout=15)
    r = h.get(
        server.base_url + "/v1/states",
        headers={"Accept-Encoding": "gzip"},
    )
    assert r.status_code == 200
    # httpx transparently decompresses; the header proves the wire encoding
    assert r.headers.get("content-encoding") == "gzip"
    assert len(r.json()) > 10
    h.close()


def test_client_timeout_configurable(daemon):
    """Client(timeout=) reaches the transport (operators tune this for
    long-running trigger-tag diag requests)."""
    from gpud_amd.client import Client

    core, server, client = daemon
    # 10.255.255.1 is unroutable: connect cannot complete within 150 ms,
    # making the timeout deterministic (racing a real loopback server with
    # a 1 ms budget flaked once in ~30 runs)
    c = Client("https://10.255.255.1:9", timeout=0.15)
    import httpx

    with pytest.raises((httpx.TimeoutException, httpx.TransportError)):
        c.healthz_raw() if hasattr(c, "healthz_raw") else c.get_health_states()
    c.close()
    c2 = Client(server.base_url, timeout=30)
    assert c2.healthz()
    c2.close()


def test_states_subset_filter_exact(daemon):
    """components= filter returns exactly the requested subset, nothing
    else (the control plane relies on this for targeted polls)."""
    core, server, client = daemon
    comps = client.get_components()
    subset = sorted(comps)[:5]
    st = client.get_health_states(components=subset)
    assert set(st) == set(s for s in subset if s in st)
    assert set(st) <= set(subset)
    for name, states in st.items():
        assert states and states[0].health
