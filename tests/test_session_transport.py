"""Session v1 HTTP transport against a fake control plane.

The reference tests its session streams against httptest servers
(pkg/session/session_test.go); here a FastAPI app stands in for the
control plane: the reader endpoint streams JSON request frames, the writer
endpoint collects responses.
"""

import json
import queue
import socket
import threading
import time

import pytest


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.fixture()
def fake_cp():
    """A minimal control plane: POST /api/v1/session with session_type
    reader → chunked stream of frames; writer → record body."""
    import uvicorn
    from fastapi import FastAPI, Request
    from fastapi.responses import StreamingResponse

    app = FastAPI()
    to_send: "queue.Queue[dict]" = queue.Queue()
    received: "queue.Queue[dict]" = queue.Queue()
    seen_headers = {}

    @app.post("/api/v1/session")
    async def session_ep(request: Request):
        stype = request.headers.get("session_type", "reader")
        seen_headers.update(
            {
                "machine_id": request.headers.get("machine_id", ""),
                "token": request.headers.get("token", ""),
            }
        )
        if stype == "writer":
            body = await request.json()
            received.put(body)
            return {"ok": True}

        def gen():
            deadline = time.time() + 30
            while time.time() < deadline:
                try:
                    frame = to_send.get(timeout=0.2)
                except queue.Empty:
                    continue
                if frame is None:
                    return
                yield json.dumps(frame) + "\n"

        return StreamingResponse(gen(), media_type="application/json")

    port = _free_port()
    config = uvicorn.Config(app, host="127.0.0.1", port=port, log_level="error")
    server = uvicorn.Server(config)
    t = threading.Thread(target=server.run, daemon=True)
    t.start()
    deadline = time.time() + 10
    while not server.started and time.time() < deadline:
        time.sleep(0.05)
    yield f"http://127.0.0.1:{port}", to_send, received, seen_headers
    to_send.put(None)
    server.should_exit = True
    t.join(timeout=5)


def test_v1_http_transport_roundtrip(fake_cp, monkeypatch, tmp_path):
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    monkeypatch.setenv("GPUD_AMDSMI_MOCK_GPUS", "1")
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config
    from gpud_amd.session import Session

    endpoint, to_send, received, seen_headers = fake_cp
    core = build_core(
        Config(data_dir=str(tmp_path)),
        in_memory_db=True,
        kmsg_writable=False,
        record_reboot=False,
    )
    for c in core.registry.all_components():
        if getattr(c, "run_mode", "") != "manual":
            c.trigger_check()
    try:
        s = Session(core, endpoint=endpoint, token="tk", machine_id="m9")
        s.start()
        to_send.put({"req_id": "q1", "method": "getToken", "data": {}})
        # keepalive pings may interleave; find our response
        deadline = time.time() + 20
        resp = None
        while time.time() < deadline:
            frame = received.get(timeout=15)
            if frame.get("req_id") == "q1":
                resp = frame
                break
        assert resp is not None, "no response frame received"
        assert resp["data"]["token"] == "tk"
        assert seen_headers["machine_id"] == "m9"
        s.stop()
    finally:
        core.close()


def test_reconnect_backoff_doubles_and_caps(monkeypatch, tmp_path):
    """Reference: session_reconnect.go:190-234 — exponential backoff with
    jitter, capped, reset by a healthy stream. Tested with injected
    sleep/jitter fns (the reference pattern: timeAfterFunc/jitterFunc)."""
    import os

    os.environ["GPUD_AMDSMI_MOCK"] = "1"
    from gpud_amd.bootstrap import build_core
    from gpud_amd.session.session import (
        RECONNECT_BASE,
        RECONNECT_MAX,
        Session,
    )

    core = build_core(in_memory_db=True, kmsg_writable=False, record_reboot=False)
    sleeps = []
    attempts = {"n": 0}

    def reader():
        attempts["n"] += 1
        if attempts["n"] == 4:
            # 4th attempt: one healthy frame, then the stream dies again
            yield {"req_id": "x", "method": "ping", "data": {}}
        if attempts["n"] >= 6:
            raise SystemExit  # stop the loop from inside
        raise ConnectionError("stream down")

    sent = []
    s = Session(
        core,
        endpoint="http://unused",
        open_reader=reader,
        send_response=sent.append,
        sleep_fn=sleeps.append,
        jitter_fn=lambda: 1.0,  # deterministic
    )
    try:
        s._serve_loop()
    except SystemExit:
        pass
    finally:
        core.close()
    # three failed attempts: base, 2x, 4x; healthy frame resets to base
    assert sleeps[0] == RECONNECT_BASE
    assert sleeps[1] == RECONNECT_BASE * 2
    assert sleeps[2] == RECONNECT_BASE * 4
    assert sleeps[3] == RECONNECT_BASE  # reset after the healthy frame
    assert all(x <= RECONNECT_MAX for x in sleeps)
    assert sent and sent[0]["data"].get("pong") is True
    assert s.reconnects >= 4
