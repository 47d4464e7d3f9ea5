"""Session v2 gRPC contract test + kap-mTLS manager tests."""

import base64
import queue
import time

import pytest


@pytest.fixture()
def mock_core(monkeypatch, tmp_path):
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    monkeypatch.setenv("GPUD_AMDSMI_MOCK_GPUS", "1")
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config

    cfg = Config(data_dir=str(tmp_path))
    core = build_core(cfg, in_memory_db=True, kmsg_writable=False, record_reboot=False)
    for c in core.registry.all_components():
        if getattr(c, "run_mode", "") != "manual":
            c.trigger_check()
    yield core
    core.close()


def test_v2_grpc_roundtrip(mock_core):
    """Agent connects over real gRPC; manager sends 'states', gets states."""
    from gpud_amd.session import Session
    from gpud_amd.session.v2 import V2Session, serve_fake_manager

    server, service, port = serve_fake_manager()
    try:
        dispatcher = Session(
            mock_core,
            endpoint="unused",
            open_reader=lambda: iter(()),
            send_response=lambda f: None,
        )
        agent = V2Session(
            dispatcher,
            endpoint=f"127.0.0.1:{port}",
            machine_id="machine-7",
            token="tok-7",
            wire="json",
            insecure=True,  # local plaintext test server
        )
        agent.start()
        hello = service.wait_hello(10)
        assert hello == {"machine_id": "machine-7", "token": "tok-7"}
        service.to_send.put({"req_id": "r1", "method": "states", "data": {}})
        resp = service.responses.get(timeout=10)
        assert resp["req_id"] == "r1"
        comps = [x["component"] for x in resp["data"]["states"]]
        assert "cpu" in comps
        # a second request over the same stream
        service.to_send.put({"req_id": "r2", "method": "getToken", "data": {}})
        resp = service.responses.get(timeout=10)
        assert resp["req_id"] == "r2"
        agent.stop()
    finally:
        server.stop(grace=None)


def test_kapmtls_stage_activate_rollback(tmp_path):
    from gpud_amd.pkg.kapmtls import Manager

    m = Manager(str(tmp_path / "kapmtls"))
    assert m.activate() is not None  # nothing staged
    v1 = m.stage(b"CERT1", b"KEY1", version="100")
    assert m.staged_version() == "100"
    assert m.activate() is None
    assert m.active_version() == "100"
    st = m.status()
    assert st["active_version"] == "100"
    with open(st["active_cert"], "rb") as f:
        assert f.read() == b"CERT1"
    # stage + activate v2, then roll back to v1
    m.stage(b"CERT2", b"KEY2", version="200")
    assert m.activate() is None
    assert m.active_version() == "200"
    assert m.rollback() is None
    assert m.active_version() == "100"


def test_kapmtls_session_methods(mock_core):
    from gpud_amd.session import Session

    s = Session(
        mock_core,
        endpoint="unused",
        open_reader=lambda: iter(()),
        send_response=lambda f: None,
    )
    resp = s.process_request(
        {
            "req_id": "k1",
            "method": "updateKAPMTLSCredentials",
            "data": {
                "cert": base64.b64encode(b"C").decode(),
                "key": base64.b64encode(b"K").decode(),
                "version": "5",
            },
        }
    )
    assert resp["data"]["staged_version"] == "5"
    resp = s.process_request({"req_id": "k2", "method": "activateKAPMTLS", "data": {}})
    assert resp["data"]["active_version"] == "5"
    resp = s.process_request({"req_id": "k3", "method": "kapMTLSStatus", "data": {}})
    assert resp["data"]["active_version"] == "5"
    resp = s.process_request({"req_id": "k4", "method": "nodeCredentials", "data": {}})
    assert "no files" in resp["data"]["error"]  # reference semantics


def test_kapmtls_grpc_credentials(tmp_path):
    """Activated credentials build real gRPC mTLS channel credentials."""
    import subprocess

    from gpud_amd.pkg.kapmtls import Manager

    d = tmp_path / "certs"
    d.mkdir()
    subprocess.run(
        [
            "openssl", "req", "-x509", "-newkey", "rsa:2048",
            "-keyout", str(d / "k.pem"), "-out", str(d / "c.pem"),
            "-days", "1", "-nodes", "-subj", "/CN=agent",
        ],
        check=True,
        capture_output=True,
    )
    m = Manager(str(tmp_path / "kap"))
    m.stage((d / "c.pem").read_bytes(), (d / "k.pem").read_bytes(), "1")
    assert m.grpc_channel_credentials() is None  # not active yet
    assert m.activate() is None
    creds = m.grpc_channel_credentials()
    import grpc

    assert isinstance(creds, grpc.ChannelCredentials)


def test_v2_protobuf_wire_contract(mock_core):
    """Full protobuf-framed contract: the agent speaks the reference's
    session.proto wire format over /gpud.session.v2.SessionService/Connect —
    Hello, hello_ack, request oneofs, Result{request_id, payload_json}."""
    from gpud_amd import __version__
    from gpud_amd.session import Session
    from gpud_amd.session.v2 import V2Session, serve_fake_manager

    server, service, port = serve_fake_manager(wire="proto")
    try:
        dispatcher = Session(
            mock_core,
            endpoint="unused",
            open_reader=lambda: iter(()),
            send_response=lambda f: None,
        )
        agent = V2Session(dispatcher, endpoint=f"127.0.0.1:{port}", insecure=True)
        assert agent.wire == "proto"  # the default is the reference framing
        agent.start()
        hello = service.wait_hello(10)
        assert hello["agent_version"] == __version__
        assert hello["min_protocol_revision"] == 1

        service.to_send.put(
            {"hello_ack": {"protocol_revision": 1, "manager_instance_id": "m1"}}
        )
        service.to_send.put({"request_id": "p1", "get_health_states": {}})
        resp = service.responses.get(timeout=10)
        assert resp["req_id"] == "p1"
        comps = [x["component"] for x in resp["data"]["states"]]
        assert "cpu" in comps

        service.to_send.put(
            {
                "request_id": "p2",
                "update_config": {"values": {"poll_interval_seconds": "30"}},
            }
        )
        resp = service.responses.get(timeout=10)
        assert resp["req_id"] == "p2"
        assert "poll_interval_seconds" in resp["data"]["applied"]

        service.to_send.put(
            {
                "request_id": "p3",
                "trigger_component": {"component_name": "cpu"},
            }
        )
        resp = service.responses.get(timeout=10)
        assert resp["req_id"] == "p3"
        assert resp["data"]["states"][0]["health"] in ("Healthy", "Degraded")

        service.to_send.put(
            {
                "request_id": "p4",
                "set_plugin_specs": {
                    "specs_present": True,
                    "specs": [
                        {
                            "plugin_name": "wired",
                            "plugin_type": "component",
                            "run_mode": "manual",
                            "health_state_plugin": {
                                "steps": [
                                    {
                                        "name": "s",
                                        "run_bash_script": {
                                            "content_type": "plaintext",
                                            "script": "echo from-proto",
                                        },
                                    }
                                ]
                            },
                            "timeout_nanos": 30_000_000_000,
                        }
                    ],
                },
            }
        )
        resp = service.responses.get(timeout=10)
        assert resp["req_id"] == "p4"
        assert "custom-plugin-wired" in resp["data"]["registered"]
        agent.stop()
    finally:
        server.stop(grace=None)


def test_daemon_boot_with_v2_session(tmp_path):
    """`gpud run --session-protocol v2` connects the protobuf-framed gRPC
    session to a live fake manager (the reference's --session-protocol)."""
    import os
    import signal
    import socket
    import subprocess
    import sys

    from gpud_amd.session.v2 import serve_fake_manager

    server, service, grpc_port = serve_fake_manager(wire="proto")

    s = socket.socket(); s.bind(("127.0.0.1", 0))
    http_port = s.getsockname()[1]; s.close()
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = {**os.environ, "GPUD_AMDSMI_MOCK": "1", "PYTHONPATH": repo}
    proc = subprocess.Popen(
        [
            sys.executable, "-m", "gpud_amd", "run",
            "--in-memory-db", "--address", f"127.0.0.1:{http_port}",
            "--log-level", "warning",
            "--endpoint", f"127.0.0.1:{grpc_port}",
            "--session-protocol", "v2",
            # the fake manager is a local plaintext gRPC server
            "--control-plane-insecure-tls",
        ],
        cwd=repo, env=env,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        start_new_session=True,
    )
    try:
        hello = service.wait_hello(30)
        assert hello is not None, "agent never connected over v2"
        assert hello["min_protocol_revision"] == 1
        service.to_send.put({"request_id": "b1", "get_health_states": {}})
        resp = service.responses.get(timeout=15)
        assert resp["req_id"] == "b1"
        assert any(x["component"] == "cpu" for x in resp["data"]["states"])
    finally:
        try:
            os.killpg(proc.pid, signal.SIGTERM)
        except ProcessLookupError:
            pass
        proc.wait(timeout=15)
        server.stop(grace=None)


def test_kapmtls_gateway_metadata_travels_with_version(tmp_path):
    from gpud_amd.pkg.kapmtls import Manager

    m = Manager(str(tmp_path / "kap"))
    m.stage(
        b"C1", b"K1", version="1",
        gateway_ca_pem=b"CA1", gateway_endpoint="gw1:443", server_name="gw1",
    )
    assert m.activate() is None
    assert m.gateway_info() == {"endpoint": "gw1:443", "server_name": "gw1"}
    m.stage(b"C2", b"K2", version="2", gateway_endpoint="gw2:443")
    assert m.activate() is None
    assert m.gateway_info()["endpoint"] == "gw2:443"
    assert m.rollback() is None
    assert m.gateway_info()["endpoint"] == "gw1:443"


def test_node_credentials_placement(mock_core, tmp_path):
    """Reference pkg/session/node_credentials.go: allow-listed paths,
    validate-all-before-write-any, atomic publish with owner-only mode."""
    import base64 as b64
    import os

    from gpud_amd.session import Session

    s = Session(
        mock_core,
        endpoint="unused",
        open_reader=lambda: iter(()),
        send_response=lambda f: None,
    )
    allowed = str(tmp_path / "allowed") + "/"
    s.NODE_CREDENTIAL_ALLOWED_PREFIXES = [allowed]

    def req(files):
        return s.process_request(
            {"req_id": "n", "method": "nodeCredentials",
             "data": {"kubelet": files}}
        )["data"]

    enc = lambda b: b64.b64encode(b).decode()  # noqa: E731
    # outside the allow-list: refused, nothing written
    out = req({"config": {"path": "/etc/passwd", "contents": enc(b"x")}})
    assert "outside" in out["error"]
    # '..' escape is normalized before the prefix check
    sneaky = allowed + "../escape"
    out = req({"config": {"path": sneaky, "contents": enc(b"x")}})
    assert "outside" in out["error"]
    # empty contents refused (a truncated credential must not look written)
    out = req({"config": {"path": allowed + "kubeconfig", "contents": ""}})
    assert "no contents" in out["error"]
    # one bad file in a pair: NOTHING is written
    good = {"path": allowed + "kubeconfig", "contents": enc(b"KC")}
    bad = {"path": "/etc/shadow", "contents": enc(b"x")}
    out = req({"config": good, "client_certificate": bad})
    assert "outside" in out["error"]
    assert not os.path.exists(good["path"])
    # happy path: both written, default mode 0600
    cert = {"path": allowed + "pki/kubelet.pem", "contents": enc(b"CERT"),
            "mode": 0o640}
    out = req({"config": good, "client_certificate": cert})
    assert out["written"] == [good["path"], cert["path"]]
    assert open(good["path"], "rb").read() == b"KC"
    assert oct(os.stat(good["path"]).st_mode & 0o777) == "0o600"
    assert oct(os.stat(cert["path"]).st_mode & 0o777) == "0o640"


def test_v2_auto_fallback_on_unimplemented(mock_core):
    """A gRPC server without the Connect method (UNIMPLEMENTED) triggers
    the one-shot legacy fallback (reference ProtocolAuto)."""
    import threading
    from concurrent.futures import ThreadPoolExecutor

    import grpc

    from gpud_amd.session import Session
    from gpud_amd.session.v2 import V2Session

    # a real gRPC server with NO handlers -> UNIMPLEMENTED for Connect
    server = grpc.server(ThreadPoolExecutor(max_workers=2))
    port = server.add_insecure_port("127.0.0.1:0")
    server.start()
    fell_back = threading.Event()
    try:
        dispatcher = Session(
            mock_core,
            endpoint="unused",
            open_reader=lambda: iter(()),
            send_response=lambda f: None,
        )
        agent = V2Session(
            dispatcher,
            endpoint=f"127.0.0.1:{port}",
            on_unsupported=fell_back.set,
            insecure=True,
        )
        agent.start()
        assert fell_back.wait(15), "fallback was not triggered"
        agent.stop()
    finally:
        server.stop(grace=None)


def test_v2_reconnects_after_manager_death(mock_core):
    """A dead manager stream triggers backoff reconnect attempts, and the
    agent thread survives to keep trying (reference keepAliveV2 loop)."""
    import time

    from gpud_amd.session import Session
    from gpud_amd.session.v2 import V2Session, serve_fake_manager

    server, service, port = serve_fake_manager(wire="proto")
    dispatcher = Session(
        mock_core,
        endpoint="unused",
        open_reader=lambda: iter(()),
        send_response=lambda f: None,
    )
    agent = V2Session(dispatcher, endpoint=f"127.0.0.1:{port}", insecure=True)
    agent.start()
    try:
        assert service.wait_hello(15)
        service.to_send.put({"request_id": "a", "get_health_states": {}})
        assert service.responses.get(timeout=15)["req_id"] == "a"
        server.stop(grace=None)
        deadline = time.time() + 30
        while agent.reconnects == 0 and time.time() < deadline:
            time.sleep(0.2)
        assert agent.reconnects >= 1
        assert agent._thread.is_alive()
    finally:
        agent.stop()
        server.stop(grace=None)
