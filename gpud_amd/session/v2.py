"""Session v2: gRPC bidirectional stream to the control plane.

Reference: pkg/session/v2/session.proto:12-53 — ``SessionService.Connect
(stream AgentPacket) returns (stream ManagerPacket)``, a protobuf mirror of
the v1 method set, selected via ``--session-protocol`` (reference:
cmd/gpud/run/command.go:156). This environment has grpcio but no protobuf
codegen toolchain, so packets are JSON-encoded bytes over a real gRPC
stream-stream method (generic handlers, identity serializers); the method
set maps 1:1 onto the v1 dispatcher like the reference's v2 adapter
(pkg/session/session_v2_adapter.go:72).

Packet shapes:
  manager -> agent: {"req_id": str, "method": str, "data": {...}}
  agent -> manager: {"req_id": str, "method": str, "data": {...}}
The first agent packet is a hello: {"hello": {"machine_id", "token"}}.
"""

from __future__ import annotations

import json
import queue
import threading
import time
from typing import Iterator, Optional

import grpc

from ..pkg.log import logger
from .session import Session

SERVICE_METHOD = "/gpud.v2.SessionService/Connect"


def _ser(obj: dict) -> bytes:
    return json.dumps(obj).encode()


def _deser(b: bytes) -> dict:
    return json.loads(b.decode())


class V2Session:
    """Agent side of the v2 gRPC session; delegates dispatch to Session."""

    def __init__(
        self,
        dispatcher: Session,
        endpoint: str,
        machine_id: str = "",
        token: str = "",
        credentials: Optional[grpc.ChannelCredentials] = None,
    ):
        self.dispatcher = dispatcher
        self.endpoint = endpoint
        self.machine_id = machine_id
        self.token = token
        self.credentials = credentials
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._outbox: "queue.Queue[Optional[dict]]" = queue.Queue()
        self.reconnects = 0

    def _request_iter(self) -> Iterator[bytes]:
        yield _ser({"hello": {"machine_id": self.machine_id, "token": self.token}})
        while not self._stop.is_set():
            try:
                item = self._outbox.get(timeout=0.5)
            except queue.Empty:
                continue
            if item is None:
                return
            yield _ser(item)

    def _run_once(self) -> None:
        if self.credentials is not None:
            channel = grpc.secure_channel(self.endpoint, self.credentials)
        else:
            channel = grpc.insecure_channel(self.endpoint)
        try:
            callable_ = channel.stream_stream(
                SERVICE_METHOD,
                request_serializer=lambda b: b,
                response_deserializer=lambda b: b,
            )
            for raw in callable_(self._request_iter()):
                if self._stop.is_set():
                    return
                try:
                    frame = _deser(raw)
                except (ValueError, UnicodeDecodeError):
                    continue
                resp = self.dispatcher.process_request(frame)
                if resp is not None:
                    self._outbox.put(resp)
        finally:
            channel.close()

    def start(self) -> None:
        self._thread = threading.Thread(
            target=self._loop, daemon=True, name="gpud-session-v2"
        )
        self._thread.start()

    def _loop(self) -> None:
        backoff = 1.0
        while not self._stop.is_set():
            try:
                self._run_once()
                backoff = 1.0
            except grpc.RpcError as e:
                logger.warning("v2 session stream error: %s", e)
            except Exception:
                logger.exception("v2 session failure")
            if self._stop.is_set():
                return
            self.reconnects += 1
            time.sleep(min(backoff, 60.0))
            backoff = min(backoff * 2, 60.0)

    def stop(self) -> None:
        self._stop.set()
        self._outbox.put(None)


# ---------------------------------------------------------------------------
# Test/reference control-plane server (the contract-test peer; reference:
# pkg/session/v2/contract_test.go)
# ---------------------------------------------------------------------------

class FakeManagerService:
    """In-process gRPC 'control plane' for contract tests: sends queued
    requests to connected agents and records their responses."""

    def __init__(self):
        self.to_send: "queue.Queue[dict]" = queue.Queue()
        self.responses: "queue.Queue[dict]" = queue.Queue()
        self.hello: Optional[dict] = None
        self._hello_evt = threading.Event()

    def handler(self, request_iterator, context):
        def reader():
            for raw in request_iterator:
                frame = _deser(raw)
                if "hello" in frame:
                    self.hello = frame["hello"]
                    self._hello_evt.set()
                else:
                    self.responses.put(frame)

        t = threading.Thread(target=reader, daemon=True)
        t.start()
        while context.is_active():
            try:
                req = self.to_send.get(timeout=0.2)
            except queue.Empty:
                continue
            if req is None:
                return
            yield _ser(req)

    def wait_hello(self, timeout: float = 10.0) -> Optional[dict]:
        self._hello_evt.wait(timeout)
        return self.hello


def serve_fake_manager(port: int = 0):
    """Returns (server, service, bound_port)."""
    service = FakeManagerService()

    method_handlers = {
        "Connect": grpc.stream_stream_rpc_method_handler(
            service.handler,
            request_deserializer=lambda b: b,
            response_serializer=lambda b: b,
        )
    }
    generic = grpc.method_handlers_generic_handler(
        "gpud.v2.SessionService", method_handlers
    )
    server = grpc.server(
        __import__("concurrent.futures", fromlist=["ThreadPoolExecutor"])
        .ThreadPoolExecutor(max_workers=4)
    )
    server.add_generic_rpc_handlers((generic,))
    bound = server.add_insecure_port(f"127.0.0.1:{port}")
    server.start()
    return server, service, bound
