"""Unit-depth pass: wire types aggregates, server error paths, runner lock,
session updateConfig, distsign negatives, kapmtls errors, log rotation."""

import datetime
import logging
import os
import threading
import time

import pytest

from gpud_amd.apiv1.types import (
    ComponentEvents,
    ComponentMetrics,
    Event,
    Info,
    Metric,
    utcnow,
)


def test_aggregate_wire_types():
    now = utcnow()
    ce = ComponentEvents(
        component="cpu",
        start_time=now - datetime.timedelta(hours=1),
        end_time=now,
        events=[Event(time=now, name="e", type="Info", message="m")],
    )
    d = ce.to_dict()
    assert d["component"] == "cpu"
    assert d["events"][0]["name"] == "e"
    assert d["startTime"].endswith("Z")
    cm = ComponentMetrics(
        component="cpu", metrics=[Metric(unix_seconds=1, name="m", value=2.0)]
    )
    assert cm.to_dict()["metrics"][0]["value"] == 2.0
    info = Info(events=[], states=[], metrics=[])
    assert info.to_dict() == {"states": [], "events": [], "metrics": []}


def test_runner_serializes(monkeypatch):
    from gpud_amd.pkg.process_runner import Runner

    r = Runner()
    results = []

    def long_job():
        results.append(r.run_until_completion("sleep 0.5; echo a", timeout_seconds=5))

    t = threading.Thread(target=long_job)
    t.start()
    time.sleep(0.1)
    # second run must wait (or time out on the lock)
    res = r.run_until_completion("echo b", timeout_seconds=0.1)
    assert res.exit_code == -1 and "another script" in res.error
    t.join()
    assert results[0].exit_code == 0


def test_distsign_malformed_inputs():
    from gpud_amd.pkg import distsign

    seed, pub = distsign.generate_keypair(b"\x09" * 32)
    sig = distsign.sign(b"msg", seed)
    assert distsign.verify(b"msg", sig, pub)
    assert not distsign.verify(b"msg", sig[:-1], pub)  # short sig
    assert not distsign.verify(b"msg", b"\x00" * 64, pub)  # garbage sig
    assert not distsign.verify(b"msg", sig, b"\x00" * 31)  # short key


def test_kapmtls_incomplete_stage(tmp_path):
    from gpud_amd.pkg.kapmtls import Manager

    m = Manager(str(tmp_path))
    # stage dir exists but missing the key file
    d = tmp_path / "v-7"
    d.mkdir()
    (d / "client.crt").write_bytes(b"C")
    (tmp_path / "staged").write_text("7")
    err = m.activate()
    assert err is not None and "incomplete" in err
    assert m.rollback() is not None  # nothing recorded


def test_log_setup_and_rotation(tmp_path):
    from gpud_amd.pkg.log import audit_logger, setup

    logfile = tmp_path / "gpud.log"
    lg = setup(level="debug", log_file=str(logfile))
    lg.info("hello log")
    for h in lg.handlers:
        h.flush()
    assert "hello log" in logfile.read_text()
    assert isinstance(audit_logger(), logging.Logger)
    setup(level="info")  # restore


def test_session_update_config(monkeypatch, tmp_path):
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config
    from gpud_amd.session import Session

    core = build_core(
        Config(data_dir=str(tmp_path)),
        in_memory_db=True,
        kmsg_writable=False,
        record_reboot=False,
    )
    try:
        s = Session(
            core,
            endpoint="unused",
            open_reader=lambda: iter(()),
            send_response=lambda f: None,
        )
        resp = s.process_request(
            {
                "req_id": "u",
                "method": "updateConfig",
                "data": {"expected_gpu_count": 8, "not_a_field": 1},
            }
        )
        assert resp["data"]["applied"] == ["expected_gpu_count"]
        assert core.config.expected_gpu_count == 8
    finally:
        core.close()


def test_server_bad_requests(monkeypatch, tmp_path):
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    import httpx

    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config
    from gpud_amd.server import Server

    core = build_core(
        Config(data_dir=str(tmp_path)),
        in_memory_db=True,
        kmsg_writable=False,
        record_reboot=False,
    )
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    server = Server(core, port=port, tls=False)
    server.start()
    try:
        base = server.base_url
        # trigger-check needs exactly one selector
        r = httpx.get(base + "/v1/components/trigger-check")
        assert r.status_code == 400
        r = httpx.get(
            base + "/v1/components/trigger-check",
            params={"componentName": "cpu", "tagName": "x"},
        )
        assert r.status_code == 400
        r = httpx.get(
            base + "/v1/components/trigger-check",
            params={"componentName": "ghost"},
        )
        assert r.status_code == 404
        # bad inject-fault body
        r = httpx.post(base + "/inject-fault", content=b"not json")
        assert r.status_code == 400
        # events honors startTime filter without error
        r = httpx.get(
            base + "/v1/events",
            params={"components": "os", "startTime": "2020-01-01T00:00:00Z"},
        )
        assert r.status_code == 200
        assert r.json()[0]["component"] == "os"
    finally:
        server.stop()
        core.close()


def test_host_boot_helpers():
    from gpud_amd.pkg import host

    bt = host.boot_time()
    assert bt < utcnow()
    assert (utcnow() - bt).total_seconds() == pytest.approx(
        host.uptime_seconds(), rel=0.1
    )
