"""Session v2: gRPC bidirectional stream to the control plane.

Reference: pkg/session/v2/session.proto:12-53 — ``SessionService.Connect
(stream AgentPacket) returns (stream ManagerPacket)``. Two wire modes:

- ``wire="proto"`` (default): the reference's actual protobuf wire format,
  hand-encoded by ``protowire.py`` (no protoc in this environment) over
  the reference's method path ``/gpud.session.v2.SessionService/Connect``.
  The agent sends ``Hello`` then ``Result{request_id, payload_json}``
  packets; decoded ManagerPacket oneofs map 1:1 onto the v1 dispatcher
  exactly like the reference's v2 adapter
  (pkg/session/session_v2_adapter.go:72).
- ``wire="json"``: the plain JSON-bytes framing kept for debugging and
  transport tests; packet shapes:
  manager -> agent: {"req_id": str, "method": str, "data": {...}}
  agent -> manager: {"req_id": str, "method": str, "data": {...}}
  with an initial {"hello": {"machine_id", "token"}} agent packet.
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

SERVICE_METHOD_JSON = "/gpud.v2.SessionService/Connect"
# the reference's generated service path (session.proto package gpud.session.v2)
SERVICE_METHOD_PROTO = "/gpud.session.v2.SessionService/Connect"


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
        wire: str = "proto",
        on_unsupported=None,
        insecure: bool = False,
    ):
        self.dispatcher = dispatcher
        self.endpoint = endpoint
        self.machine_id = machine_id
        self.token = token
        # TLS by default (reference: session_v2.go:278 — verified TLS, no
        # InsecureSkipVerify); a plaintext channel needs the explicit
        # insecure flag (tests, lab setups)
        if credentials is None and not insecure:
            credentials = grpc.ssl_channel_credentials()
        self.credentials = credentials
        self.wire = wire
        # "auto" protocol support (reference: session_keepalive.go:15 —
        # ProtocolAuto tries v2 and falls back to the legacy session when
        # the manager reports it unsupported): called once on UNIMPLEMENTED
        self.on_unsupported = on_unsupported
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._outbox: "queue.Queue[Optional[dict]]" = queue.Queue()
        self.reconnects = 0

    def _request_iter(self) -> Iterator[bytes]:
        if self.wire == "proto":
            from .. import __version__
            from . import protowire as pw

            yield pw.hello_bytes(agent_version=__version__)
        else:
            yield _ser(
                {"hello": {"machine_id": self.machine_id, "token": self.token}}
            )
        while not self._stop.is_set():
            try:
                item = self._outbox.get(timeout=0.5)
            except queue.Empty:
                continue
            if item is None:
                return
            yield item if isinstance(item, bytes) else _ser(item)

    def _run_once(self) -> None:
        if self.credentials is not None:
            channel = grpc.secure_channel(self.endpoint, self.credentials)
        else:
            channel = grpc.insecure_channel(self.endpoint)
        try:
            callable_ = channel.stream_stream(
                SERVICE_METHOD_PROTO if self.wire == "proto" else SERVICE_METHOD_JSON,
                request_serializer=lambda b: b,
                response_deserializer=lambda b: b,
            )
            for raw in callable_(self._request_iter()):
                if self._stop.is_set():
                    return
                if self.wire == "proto":
                    self._handle_proto_packet(raw)
                    continue
                try:
                    frame = _deser(raw)
                except (ValueError, UnicodeDecodeError):
                    continue
                resp = self.dispatcher.process_request(frame)
                if resp is not None:
                    self._outbox.put(resp)
        finally:
            channel.close()

    def _handle_proto_packet(self, raw: bytes) -> None:
        from . import protowire as pw

        try:
            frame = pw.manager_packet_to_frame(raw)
        except ValueError:
            logger.warning("v2: undecodable ManagerPacket (%d bytes)", len(raw))
            return
        if frame is None:
            return
        if frame.get("_control") == "hello_ack":
            logger.info(
                "v2 session established: manager %s revision %s",
                frame.get("manager_instance_id", "?"),
                frame.get("protocol_revision", "?"),
            )
            return
        if frame.get("_control") == "drain_notice":
            delay = frame.get("reconnect_after_millis", 0) / 1000.0
            logger.info("v2 drain notice: reconnect after %.1fs", delay)
            time.sleep(min(delay, 60.0))
            raise grpc.RpcError()  # leave the stream; the loop reconnects
        resp = self.dispatcher.process_request(frame)
        if resp is not None:
            self._outbox.put(pw.result_bytes(resp.get("req_id", ""), resp.get("data")))

    def start(self) -> None:
        self._thread = threading.Thread(
            target=self._loop, daemon=True, name="gpud-session-v2"
        )
        self._thread.start()

    def _loop(self) -> None:
        backoff = 1.0
        consecutive_failures = 0
        while not self._stop.is_set():
            try:
                self._run_once()
                backoff = 1.0
                consecutive_failures = 0
            except grpc.RpcError as e:
                code = e.code() if hasattr(e, "code") else None
                consecutive_failures += 1
                # UNIMPLEMENTED = the manager definitively lacks v2; repeated
                # UNAVAILABLE covers a v1-only HTTP endpoint that cannot even
                # speak gRPC (the reference's infra serves both on one port,
                # ours may not)
                if self.on_unsupported is not None and (
                    code == grpc.StatusCode.UNIMPLEMENTED
                    or consecutive_failures >= 3
                ):
                    logger.info(
                        "v2 session unsupported by the manager; "
                        "falling back to the legacy session"
                    )
                    cb, self.on_unsupported = self.on_unsupported, None
                    self._stop.set()
                    cb()
                    return
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
    requests to connected agents and records their responses.

    ``wire="proto"`` speaks the reference protobuf framing: queue
    ManagerPacket dicts on ``to_send``; ``responses`` receives decoded
    Result dicts {"req_id", "data"}; ``hello`` is the decoded Hello."""

    def __init__(self, wire: str = "json"):
        self.wire = wire
        self.to_send: "queue.Queue[dict]" = queue.Queue()
        self.responses: "queue.Queue[dict]" = queue.Queue()
        self.hello: Optional[dict] = None
        self._hello_evt = threading.Event()

    def handler(self, request_iterator, context):
        def reader():
            try:
                self._read_stream(request_iterator)
            except grpc.RpcError:
                return  # agent hung up; normal at teardown

        t = threading.Thread(target=reader, daemon=True)
        t.start()
        while context.is_active():
            try:
                req = self.to_send.get(timeout=0.2)
            except queue.Empty:
                continue
            if req is None:
                return
            if self.wire == "proto":
                from . import protowire as pw

                yield pw.encode_message("ManagerPacket", req)
            else:
                yield _ser(req)

    def _read_stream(self, request_iterator):
            for raw in request_iterator:
                if self.wire == "proto":
                    from . import protowire as pw

                    pkt = pw.decode_message("AgentPacket", raw)
                    if "hello" in pkt:
                        self.hello = pkt["hello"]
                        self._hello_evt.set()
                    elif "result" in pkt:
                        res = pkt["result"]
                        self.responses.put(
                            {
                                "req_id": res.get("request_id", ""),
                                "data": json.loads(
                                    res.get("payload_json", b"null") or b"null"
                                ),
                            }
                        )
                    continue
                frame = _deser(raw)
                if "hello" in frame:
                    self.hello = frame["hello"]
                    self._hello_evt.set()
                else:
                    self.responses.put(frame)

    def wait_hello(self, timeout: float = 10.0) -> Optional[dict]:
        self._hello_evt.wait(timeout)
        return self.hello


def serve_fake_manager(port: int = 0, wire: str = "json"):
    """Returns (server, service, bound_port)."""
    service = FakeManagerService(wire=wire)

    method_handlers = {
        "Connect": grpc.stream_stream_rpc_method_handler(
            service.handler,
            request_deserializer=lambda b: b,
            response_serializer=lambda b: b,
        )
    }
    generic = grpc.method_handlers_generic_handler(
        "gpud.session.v2.SessionService" if wire == "proto"
        else "gpud.v2.SessionService",
        method_handlers,
    )
    server = grpc.server(
        __import__("concurrent.futures", fromlist=["ThreadPoolExecutor"])
        .ThreadPoolExecutor(max_workers=4)
    )
    server.add_generic_rpc_handlers((generic,))
    bound = server.add_insecure_port(f"127.0.0.1:{port}")
    server.start()
    return server, service, bound
