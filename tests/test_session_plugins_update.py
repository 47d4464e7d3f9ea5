"""Session dispatch, custom plugins, distsign/update, package manager."""

import base64
import io
import json
import os
import tarfile
import threading
import time

import pytest

from gpud_amd.apiv1.types import HealthStateType


@pytest.fixture()
def mock_core(monkeypatch, tmp_path):
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    monkeypatch.setenv("GPUD_AMDSMI_MOCK_GPUS", "2")
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config

    cfg = Config(data_dir=str(tmp_path))
    core = build_core(cfg, in_memory_db=True, kmsg_writable=False, record_reboot=False)
    for c in core.registry.all_components():
        if getattr(c, "run_mode", "") != "manual":
            c.trigger_check()
    yield core
    core.close()


# ---------------------------------------------------------------------------
# custom plugins
# ---------------------------------------------------------------------------

def test_plugin_spec_load_and_run(tmp_path):
    from gpud_amd.pkg import custom_plugins as cp

    specs_yaml = tmp_path / "plugins.yaml"
    specs_yaml.write_text(
        """
- plugin_name: hello
  plugin_type: component
  tags: [demo]
  timeout: 10s
  interval: 1m
  health_state_plugin:
    steps:
      - run_bash_script:
          content_type: plaintext
          script: |
            echo '{"result": "ok", "level": "warn"}'
    parser:
      json_paths:
        - query: result
          field: result
          expect:
            regex: "^ok$"
        - query: level
          field: level
          suggested_actions:
            REBOOT_SYSTEM: "panic"
"""
    )
    specs = cp.load_specs(str(specs_yaml))
    assert len(specs) == 1
    comp = cp.make_components(specs[0])[0]
    assert comp.name == "custom-plugin-hello"
    assert comp.deregisterable()
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY, cr.reason
    assert cr.extra_info["result"] == "ok"
    assert cr.suggested_actions is None  # "warn" does not match "panic"


def test_plugin_expect_failure_and_actions():
    from gpud_amd.pkg import custom_plugins as cp

    spec = cp.Spec.from_dict(
        {
            "plugin_name": "failing",
            "plugin_type": "component",
            "health_state_plugin": {
                "steps": [
                    {
                        "run_bash_script": {
                            "script": 'echo \'{"status": "panic now"}\''
                        }
                    }
                ],
                "parser": {
                    "json_paths": [
                        {
                            "query": "status",
                            "field": "status",
                            "expect": {"regex": "^healthy$"},
                        }
                    ]
                },
            },
        }
    )
    cr = cp.make_components(spec)[0].trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert "does not match" in cr.reason


def test_plugin_step_failure():
    from gpud_amd.pkg import custom_plugins as cp

    spec = cp.Spec.from_dict(
        {
            "plugin_name": "bad",
            "plugin_type": "component",
            "health_state_plugin": {
                "steps": [{"run_bash_script": {"script": "exit 7"}}]
            },
        }
    )
    cr = cp.make_components(spec)[0].trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert "exited 7" in cr.reason


def test_plugin_base64_and_component_list():
    from gpud_amd.pkg import custom_plugins as cp

    script = base64.b64encode(b"echo item ${NAME}").decode()
    spec = cp.Spec.from_dict(
        {
            "plugin_name": "multi",
            "plugin_type": "component_list",
            "component_list": ["a", "b"],
            "health_state_plugin": {
                "steps": [
                    {
                        "run_bash_script": {
                            "content_type": "base64",
                            "script": script,
                        }
                    }
                ]
            },
        }
    )
    comps = cp.make_components(spec)
    assert [c.name for c in comps] == [
        "custom-plugin-multi-a",
        "custom-plugin-multi-b",
    ]
    cr = comps[1].trigger_check()
    assert cr.health == HealthStateType.HEALTHY
    assert "item b" in cr.raw_output


def test_init_plugin_gates_start():
    from gpud_amd.pkg import custom_plugins as cp

    ok = cp.Spec.from_dict(
        {
            "plugin_name": "init-ok",
            "plugin_type": "init",
            "health_state_plugin": {"steps": [{"run_bash_script": {"script": "true"}}]},
        }
    )
    bad = cp.Spec.from_dict(
        {
            "plugin_name": "init-bad",
            "plugin_type": "init",
            "health_state_plugin": {"steps": [{"run_bash_script": {"script": "false"}}]},
        }
    )
    assert cp.run_init_plugins([ok]) is None
    err = cp.run_init_plugins([ok, bad])
    assert err is not None and "init-bad" in err


def test_parse_duration():
    from gpud_amd.pkg.custom_plugins import parse_duration

    assert parse_duration("90s") == 90
    assert parse_duration("10m") == 600
    assert parse_duration("1h") == 3600
    assert parse_duration(42) == 42


# ---------------------------------------------------------------------------
# session dispatch
# ---------------------------------------------------------------------------

def _session(core):
    from gpud_amd.session import Session

    return Session(
        core,
        endpoint="https://cp.example",
        token="tok",
        machine_id="m1",
        open_reader=lambda: iter(()),
        send_response=lambda frame: None,
    )


def test_session_states_events_metrics(mock_core):
    s = _session(mock_core)
    resp = s.process_request({"req_id": "1", "method": "states", "data": {}})
    assert resp["req_id"] == "1"
    comps = [x["component"] for x in resp["data"]["states"]]
    assert "accelerator-amd-temperature" in comps
    resp = s.process_request({"req_id": "2", "method": "events", "data": {}})
    assert "events" in resp["data"]
    resp = s.process_request({"req_id": "3", "method": "metrics", "data": {}})
    assert "metrics" in resp["data"]


def test_session_trigger_and_sethealthy(mock_core):
    s = _session(mock_core)
    resp = s.process_request(
        {
            "req_id": "4",
            "method": "triggerComponentCheck",
            "data": {"componentName": "cpu"},
        }
    )
    assert resp["data"]["states"][0]["health"] == "Healthy"
    resp = s.process_request(
        {
            "req_id": "5",
            "method": "setHealthy",
            "data": {"components": ["accelerator-amd-error-ras"]},
        }
    )
    assert resp["data"]["set_healthy"] == ["accelerator-amd-error-ras"]


def test_session_inject_fault(mock_core):
    s = _session(mock_core)
    resp = s.process_request(
        {
            "req_id": "6",
            "method": "injectFault",
            "data": {"ras_event_name": "amdgpu_gpu_reset_begin"},
        }
    )
    assert resp["data"]["status"] == "injected"


def test_session_plugin_specs_roundtrip(mock_core):
    s = _session(mock_core)
    spec_dict = {
        "plugin_name": "sess-plugin",
        "plugin_type": "component",
        "health_state_plugin": {"steps": [{"run_bash_script": {"script": "true"}}]},
    }
    resp = s.process_request(
        {"req_id": "7", "method": "setPluginSpecs", "data": {"specs": [spec_dict]}}
    )
    assert resp["data"]["registered"] == ["custom-plugin-sess-plugin"]
    assert mock_core.registry.get("custom-plugin-sess-plugin") is not None
    resp = s.process_request({"req_id": "8", "method": "getPluginSpecs", "data": {}})
    assert resp["data"]["specs"][0]["plugin_name"] == "sess-plugin"
    # plugins are deregisterable through the session
    resp = s.process_request(
        {
            "req_id": "9",
            "method": "deregisterComponent",
            "data": {"componentName": "custom-plugin-sess-plugin"},
        }
    )
    assert resp["data"]["deregistered"] == "custom-plugin-sess-plugin"


def test_session_bootstrap_and_gossip(mock_core):
    s = _session(mock_core)
    script = base64.b64encode(b"echo bootstrapped").decode()
    resp = s.process_request(
        {"req_id": "a", "method": "bootstrap", "data": {"script": script}}
    )
    assert resp["data"]["exit_code"] == 0
    assert "bootstrapped" in resp["data"]["output"]
    resp = s.process_request({"req_id": "b", "method": "gossip", "data": {}})
    assert resp["data"]["machineInfo"]["hostname"]


def test_session_token_and_logout(mock_core):
    from gpud_amd.pkg import metadata

    s = _session(mock_core)
    resp = s.process_request(
        {"req_id": "c", "method": "updateToken", "data": {"token": "newtok"}}
    )
    assert resp["data"]["status"] == "token updated"
    assert metadata.get_value(mock_core.db_ro, metadata.KEY_TOKEN) == "newtok"
    resp = s.process_request({"req_id": "d", "method": "logout", "data": {}})
    assert metadata.get_value(mock_core.db_ro, metadata.KEY_TOKEN) == ""


def test_session_unknown_method(mock_core):
    s = _session(mock_core)
    resp = s.process_request({"req_id": "e", "method": "bogus", "data": {}})
    assert "unknown method" in resp["data"]["error"]


def test_session_reconnect_backoff(mock_core):
    from gpud_amd.session import Session

    sleeps = []
    attempts = {"n": 0}

    def failing_reader():
        attempts["n"] += 1
        if attempts["n"] >= 4:
            raise StopIteration  # will surface as RuntimeError in generator
        raise ConnectionError("stream down")

    s = Session(
        mock_core,
        endpoint="https://cp.example",
        open_reader=failing_reader,
        send_response=lambda f: None,
        sleep_fn=lambda t: sleeps.append(t),
        jitter_fn=lambda: 1.0,
    )
    t = threading.Thread(target=s._serve_loop, daemon=True)
    t.start()
    deadline = time.time() + 5
    while attempts["n"] < 3 and time.time() < deadline:
        time.sleep(0.01)
    s.stop()
    t.join(timeout=2)
    assert len(sleeps) >= 2
    assert sleeps[1] >= sleeps[0]  # exponential growth


# ---------------------------------------------------------------------------
# distsign + update + package manager
# ---------------------------------------------------------------------------

def test_distsign_sign_verify_chain():
    from gpud_amd.pkg import distsign

    root_seed, root_pub = distsign.generate_keypair(b"\x01" * 32)
    sign_seed, sign_pub = distsign.generate_keypair(b"\x02" * 32)
    artifact = b"the release bytes"
    spub_sig = distsign.sign(sign_pub, root_seed)
    art_sig = distsign.sign(artifact, sign_seed)
    assert distsign.verify_release(artifact, art_sig, sign_pub, spub_sig, root_pub)
    # tampered artifact fails
    assert not distsign.verify_release(
        artifact + b"x", art_sig, sign_pub, spub_sig, root_pub
    )
    # unsanctioned signing key fails
    rogue_seed, rogue_pub = distsign.generate_keypair(b"\x03" * 32)
    rogue_sig = distsign.sign(artifact, rogue_seed)
    assert not distsign.verify_release(
        artifact, rogue_sig, rogue_pub, spub_sig, root_pub
    )


def test_update_from_local_file(tmp_path):
    from gpud_amd.pkg import distsign
    from gpud_amd.pkg.config import Config
    from gpud_amd.pkg.update import read_target_version, update_to_version

    # build a release tarball + signature chain
    payload = tmp_path / "payload"
    payload.mkdir()
    (payload / "gpud-amd").write_text("#!/bin/sh\necho v9.9.9\n")
    tar_path = tmp_path / "gpud-amd_9.9.9.tar.gz"
    with tarfile.open(tar_path, "w:gz") as tf:
        tf.add(payload / "gpud-amd", arcname="gpud-amd")
    artifact = tar_path.read_bytes()
    root_seed, root_pub = distsign.generate_keypair(b"\x07" * 32)
    sign_seed, sign_pub = distsign.generate_keypair(b"\x08" * 32)
    (tmp_path / "gpud-amd_9.9.9.tar.gz.sig").write_bytes(
        distsign.sign(artifact, sign_seed)
    )
    (tmp_path / "gpud-amd_9.9.9.tar.gz.pub").write_bytes(sign_pub)
    (tmp_path / "gpud-amd_9.9.9.tar.gz.pub.sig").write_bytes(
        distsign.sign(sign_pub, root_seed)
    )

    cfg = Config(data_dir=str(tmp_path / "data"))
    err = update_to_version(
        cfg,
        "9.9.9",
        base_url=f"file://{tmp_path}",
        root_pub=root_pub,
        install_dir=str(tmp_path / "install"),
    )
    assert err is None
    assert (tmp_path / "install" / "gpud-amd").exists()
    assert read_target_version(cfg) == "9.9.9"


def test_package_manager(tmp_path):
    from gpud_amd.pkg.config import Config
    from gpud_amd.pkg.gpud_manager import PackageController, package_statuses

    cfg = Config(data_dir=str(tmp_path))
    pkg = tmp_path / "packages" / "demo"
    pkg.mkdir(parents=True)
    marker = tmp_path / "installed-marker"
    (pkg / "init.sh").write_text(
        f"""#!/bin/bash
case "$1" in
  isInstalled) [ -f {marker} ] && exit 0 || exit 1 ;;
  install) touch {marker}; exit 0 ;;
  start) exit 0 ;;
  status) exit 0 ;;
  version) echo 1.2.3 ;;
  run) exit 0 ;;
  *) exit 1 ;;
esac
"""
    )
    sts = package_statuses(cfg)
    assert sts[0].name == "demo"
    assert sts[0].phase == "Installing"
    ctl = PackageController(cfg, interval_seconds=3600)
    ctl.reconcile_once()
    assert marker.exists()
    sts = package_statuses(cfg)
    assert sts[0].phase == "Installed"
    assert sts[0].current_version == "1.2.3"


def test_support_bundle_collect_and_session(tmp_path, monkeypatch):
    """Bundle collector (nvidia-bug-report analog) + diagnostic bundle mode."""
    import tarfile

    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    monkeypatch.setenv("GPUD_AMDSMI_MOCK_GPUS", "2")
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.bundle import collect_bundle
    from gpud_amd.pkg.config import Config
    from gpud_amd.session import Session

    core = build_core(
        Config(data_dir=str(tmp_path)),
        in_memory_db=True,
        kmsg_writable=False,
        record_reboot=False,
    )
    try:
        core.registry.get("cpu").trigger_check()
        out = str(tmp_path / "bundle.tar.gz")
        path = collect_bundle(out, core=core)
        with tarfile.open(path) as tf:
            names = tf.getnames()
            assert "gpud-bundle/bundle-info.json" in names
            assert "gpud-bundle/amdsmi.json" in names
            assert "gpud-bundle/states.json" in names
            import json as _json

            smi_doc = _json.load(tf.extractfile("gpud-bundle/amdsmi.json"))
            assert smi_doc["device_count"] == 2
            assert smi_doc["devices"]
            states = _json.load(tf.extractfile("gpud-bundle/states.json"))
            assert "cpu" in states

        s = Session(
            core,
            endpoint="unused",
            open_reader=lambda: iter(()),
            send_response=lambda f: None,
        )
        resp = s.process_request(
            {
                "req_id": "b",
                "method": "diagnostic",
                "data": {"bundle": True, "path": str(tmp_path / "b2.tar.gz")},
            }
        )
        assert resp["data"]["size"] > 0
        assert tarfile.is_tarfile(resp["data"]["bundle"])
    finally:
        core.close()


def test_managed_diagnostic_protocol(tmp_path, monkeypatch):
    """Reference docs/INTEGRATION.md: session diagnostic carries report_id +
    fixed type + timeout; runs async; uploads to the presigned URL; POSTs
    the failure endpoint when the upload fails."""
    import threading
    import time
    from http.server import BaseHTTPRequestHandler, HTTPServer

    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config
    from gpud_amd.session import Session

    hits = []

    class H(BaseHTTPRequestHandler):
        def do_PUT(self):
            self.rfile.read(int(self.headers.get("Content-Length", 0)))
            hits.append(("PUT", self.path))
            code = 500 if "fail-upload" in self.path else 200
            self.send_response(code)
            self.end_headers()

        def do_POST(self):
            self.rfile.read(int(self.headers.get("Content-Length", 0)))
            hits.append(("POST", self.path))
            self.send_response(200)
            self.end_headers()

        def log_message(self, *a):
            pass

    srv = HTTPServer(("127.0.0.1", 0), H)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    base = f"http://127.0.0.1:{srv.server_port}"

    core = build_core(
        Config(data_dir=str(tmp_path)),
        in_memory_db=True,
        kmsg_writable=False,
        record_reboot=False,
    )
    s = Session(
        core,
        endpoint=base,
        open_reader=lambda: iter(()),
        send_response=lambda f: None,
    )
    try:
        # unsupported type is refused
        resp = s.process_request(
            {"req_id": "d0", "method": "diagnostic",
             "data": {"report_id": "r0", "type": "run_my_script"}}
        )
        assert "unsupported" in resp["data"]["error"]
        # happy path: accepted, bundle uploaded
        resp = s.process_request(
            {"req_id": "d1", "method": "diagnostic",
             "data": {"report_id": "r1", "upload_url": f"{base}/up/ok"}}
        )
        assert resp["data"]["status"] == "accepted"
        for _ in range(100):
            if ("PUT", "/up/ok") in hits:
                break
            time.sleep(0.1)
        assert ("PUT", "/up/ok") in hits
        assert not any(m == "POST" for m, _ in hits)
        # failing upload triggers the failure endpoint
        s.process_request(
            {"req_id": "d2", "method": "diagnostic",
             "data": {"report_id": "r2", "upload_url": f"{base}/up/fail-upload"}}
        )
        for _ in range(100):
            if ("POST", "/api/v1/diagnostics/r2/failure") in hits:
                break
            time.sleep(0.1)
        assert ("POST", "/api/v1/diagnostics/r2/failure") in hits
    finally:
        s.stop()
        core.close()
        srv.shutdown()


def test_component_list_entry_formats(tmp_path):
    """The four entry formats + file-based lists + ${NAME}/${PAR}
    substitution (reference docs/PLUGIN.md Component List Format)."""
    from gpud_amd.pkg import custom_plugins as cp

    assert cp.parse_component_list_entry("db#manual:/var/db") == (
        "db", "manual", "/var/db",
    )
    assert cp.parse_component_list_entry("db#manual") == ("db", "manual", "")
    assert cp.parse_component_list_entry("db:/var/db") == ("db", "", "/var/db")
    assert cp.parse_component_list_entry("db") == ("db", "", "")

    listfile = tmp_path / "list.txt"
    listfile.write_text(
        "# a comment\n"
        "\n"
        "root:/\n"
        "slow#manual:/var\n"
        " #weird\n"  # indented => a real (odd) name, per the doc's escape
    )
    spec = cp.Spec.from_dict(
        {
            "plugin_name": "fmt",
            "plugin_type": "component_list",
            "component_list_file": str(listfile),
            "health_state_plugin": {
                "steps": [
                    {"run_bash_script": {
                        "content_type": "plaintext",
                        "script": "echo name=${NAME} par=${PAR}",
                    }}
                ]
            },
        }
    )
    assert spec.validate() is None
    comps = cp.make_components(spec)
    by_name = {c.name: c for c in comps}
    assert set(by_name) == {
        "custom-plugin-fmt-root",
        "custom-plugin-fmt-slow",
        "custom-plugin-fmt-#weird",
    }
    assert by_name["custom-plugin-fmt-slow"].run_mode == "manual"
    cr = by_name["custom-plugin-fmt-root"].trigger_check()
    assert "name=root par=/" in cr.raw_output
    # a list-type spec with neither list nor file is invalid
    bad = cp.Spec.from_dict(
        {
            "plugin_name": "empty",
            "plugin_type": "component_list",
            "health_state_plugin": {
                "steps": [{"run_bash_script": {"script": "true"}}]
            },
        }
    )
    assert bad.validate() is not None


def test_plugin_log_path_substitution(tmp_path):
    """log_path writes RFC3339-stamped output lines with ${PLUGIN}/${TRIGGER}
    substituted (reference docs/PLUGIN.md)."""
    from gpud_amd.pkg import custom_plugins as cp

    spec = cp.Spec.from_dict(
        {
            "plugin_name": "logger",
            "plugin_type": "component",
            "run_mode": "auto",
            "health_state_plugin": {
                "steps": [{"run_bash_script": {"script": "echo hello-log"}}],
                "parser": {
                    "log_path": str(tmp_path / "${PLUGIN}" / "${TRIGGER}.log")
                },
            },
        }
    )
    comp = cp.make_components(spec)[0]
    comp.trigger_check()
    comp.trigger_check()
    log = (tmp_path / "logger" / "auto.log").read_text()
    lines = [l for l in log.splitlines() if l]
    assert len(lines) == 2
    assert "plugin=logger trigger=auto" in lines[0]
    assert "hello-log" in lines[0]
    assert lines[0].startswith("[20")  # RFC3339 timestamp


def test_plugin_parser_extracts_embedded_json():
    """Parser finds the first valid JSON object even when embedded in other
    text (reference docs/PLUGIN.md parser note)."""
    from gpud_amd.pkg import custom_plugins as cp

    spec = cp.Spec.from_dict(
        {
            "plugin_name": "embed",
            "plugin_type": "component",
            "health_state_plugin": {
                "steps": [
                    {"run_bash_script": {
                        "script": 'echo \'prefix text {"result": "success"} suffix\''
                    }}
                ],
                "parser": {
                    "json_paths": [
                        {"query": "result", "field": "result",
                         "expect": {"regex": "^success$"}}
                    ]
                },
            },
        }
    )
    comp = cp.make_components(spec)[0]
    cr = comp.trigger_check()
    assert cr.health == "Healthy", cr.reason
    assert cr.extra_info["result"] == "success"


def test_update_pinned_root_key_enforced(tmp_path):
    """A root.pub in the data dir makes signature verification MANDATORY:
    unsigned/tampered releases are refused; a valid chain installs."""
    import io
    import os
    import tarfile

    from gpud_amd.pkg import distsign
    from gpud_amd.pkg.config import Config
    from gpud_amd.pkg.update import update_to_version

    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w:gz") as tf:
        data = b"payload"
        info = tarfile.TarInfo(name="gpud-amd/bin/gpud-amd")
        info.size = len(data)
        tf.addfile(info, io.BytesIO(data))
    art = buf.getvalue()
    (tmp_path / "gpud-amd_9.9.9.tar.gz").write_bytes(art)

    cfg = Config(data_dir=str(tmp_path / "data"))
    os.makedirs(cfg.data_dir, exist_ok=True)
    # fixed seed whose public key ENDS in a whitespace byte — a naive
    # strip() of the pinned key file corrupted it (found as a 5% flake)
    root_seed = bytes(range(32))
    for _ in range(4096):
        _, cand = distsign.generate_keypair(root_seed)
        if cand[-1:] in b" \t\n\r" or cand[:1] in b" \t\n\r":
            break
        root_seed = distsign.sign(b"next", root_seed)[:32]
    _, root_pub = distsign.generate_keypair(root_seed)
    sign_seed = os.urandom(32)
    _, sign_pub = distsign.generate_keypair(sign_seed)
    (tmp_path / "data" / "root.pub").write_bytes(root_pub)

    # pinned key + missing signature files => refused
    err = update_to_version(
        cfg, "9.9.9", base_url=f"file://{tmp_path}",
        install_dir=str(tmp_path / "i1"),
    )
    assert err is not None and "signature" in err

    # full valid chain => installs
    name = "gpud-amd_9.9.9.tar.gz"
    (tmp_path / f"{name}.sig").write_bytes(distsign.sign(art, sign_seed))
    (tmp_path / f"{name}.pub").write_bytes(sign_pub)
    (tmp_path / f"{name}.pub.sig").write_bytes(distsign.sign(sign_pub, root_seed))
    err = update_to_version(
        cfg, "9.9.9", base_url=f"file://{tmp_path}",
        install_dir=str(tmp_path / "i2"),
    )
    assert err is None
    assert (tmp_path / "i2" / "gpud-amd" / "bin" / "gpud-amd").exists()

    # tampered artifact => refused even with the old signatures present
    (tmp_path / name).write_bytes(art + b"x")
    err = update_to_version(
        cfg, "9.9.9", base_url=f"file://{tmp_path}",
        install_dir=str(tmp_path / "i3"),
    )
    assert err is not None and "FAILED" in err


def test_update_config_expected_counts_apply_live(mock_core):
    """updateConfig's expected_gpu_count / expected_xgmi_link_count reach
    the running components without a restart (reference: SetDefault*
    setters re-read per check)."""
    cfg = mock_core.config
    comp = mock_core.registry.get("accelerator-amd-gpu-counts")
    base = comp.trigger_check()
    assert base.health == "Healthy"
    cfg.expected_gpu_count = 99  # what updateConfig's setattr does
    cr = comp.trigger_check()
    assert cr.health == "Unhealthy"
    assert "expected 99" in cr.reason
    cfg.expected_gpu_count = 0
    comp.expected = 0
    assert comp.trigger_check().health == "Healthy"


def test_update_config_poll_interval_reaches_tickers(mock_core):
    sess = _session(mock_core)
    comp = mock_core.registry.get("cpu")
    before = comp.poll_interval
    out = sess._m_updateConfig({"poll_interval_seconds": 7})
    assert "poll_interval_seconds" in out["applied"]
    assert comp.poll_interval == 7.0
    sess._m_updateConfig({"poll_interval_seconds": before})
