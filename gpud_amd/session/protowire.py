"""Hand-written protobuf wire codec for the session v2 schema.

Reference: pkg/session/v2/session.proto — ``SessionService.Connect(stream
AgentPacket) returns (stream ManagerPacket)``. This environment has no
protoc/buf toolchain, so instead of shipping generated stubs the proto3
WIRE FORMAT is implemented directly (varint tags, LEN-delimited strings/
bytes/messages, map entries as repeated key/value messages). The schema
below mirrors session.proto field-for-field, so the bytes on the wire are
exactly what the reference's generated code produces/consumes.

Layering:
  - generic codec: ``encode_message(name, dict)`` / ``decode_message``
    driven by the MESSAGES descriptors;
  - adapter: ``manager_packet_to_frame`` maps a decoded ManagerPacket onto
    the v1 dispatch frame {"req_id", "method", "data"} (the reference's
    session_v2_adapter.go:72 does the same 1:1 mapping), and
    ``hello_bytes`` / ``result_bytes`` build the two AgentPacket shapes.
"""

from __future__ import annotations

import datetime
import json
from typing import Any, Dict, List, Optional, Tuple

WIRE_VARINT = 0
WIRE_I64 = 1
WIRE_LEN = 2
WIRE_I32 = 5


# ---------------------------------------------------------------------------
# low-level wire primitives
# ---------------------------------------------------------------------------


def encode_varint(n: int) -> bytes:
    if n < 0:  # proto int64: two's-complement over 64 bits, 10 bytes
        n &= (1 << 64) - 1
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def decode_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        if pos >= len(buf):
            raise ValueError("truncated varint")
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7
        if shift > 70:
            raise ValueError("varint too long")


def _tag(field_no: int, wire: int) -> bytes:
    return encode_varint((field_no << 3) | wire)


def _len_field(field_no: int, payload: bytes) -> bytes:
    return _tag(field_no, WIRE_LEN) + encode_varint(len(payload)) + payload


def _skip(buf: bytes, pos: int, wire: int) -> int:
    if wire == WIRE_VARINT:
        _, pos = decode_varint(buf, pos)
        return pos
    if wire == WIRE_I64:
        return pos + 8
    if wire == WIRE_LEN:
        ln, pos = decode_varint(buf, pos)
        return pos + ln
    if wire == WIRE_I32:
        return pos + 4
    raise ValueError(f"unsupported wire type {wire}")


def _to_signed64(n: int) -> int:
    return n - (1 << 64) if n >= (1 << 63) else n


# ---------------------------------------------------------------------------
# schema (field-for-field mirror of session.proto)
# ---------------------------------------------------------------------------
# kind: "uint32" | "int64" | "bool" | "string" | "bytes"
#       | "msg:<Name>" | "rep_string" | "rep_msg:<Name>" | "map_str_str"
#       | "map_str_msg:<Name>"

MESSAGES: Dict[str, Dict[int, Tuple[str, str]]] = {
    "AgentPacket": {1: ("hello", "msg:Hello"), 2: ("result", "msg:Result")},
    "Hello": {
        1: ("min_protocol_revision", "uint32"),
        2: ("max_protocol_revision", "uint32"),
        3: ("agent_version", "string"),
        4: ("max_receive_message_bytes", "uint32"),
        5: ("capabilities", "rep_string"),
    },
    "HelloAck": {
        1: ("protocol_revision", "uint32"),
        2: ("manager_instance_id", "string"),
        3: ("max_receive_message_bytes", "uint32"),
    },
    "Result": {1: ("request_id", "string"), 2: ("payload_json", "bytes")},
    "ManagerPacket": {
        1: ("hello_ack", "msg:HelloAck"),
        3: ("drain_notice", "msg:DrainNotice"),
        4: ("request_id", "string"),
        10: ("get_health_states", "msg:GetHealthStatesRequest"),
        11: ("get_events", "msg:GetEventsRequest"),
        12: ("get_metrics", "msg:GetMetricsRequest"),
        13: ("update", "msg:UpdateRequest"),
        14: ("set_healthy", "msg:SetHealthyRequest"),
        15: ("reboot", "msg:RebootRequest"),
        16: ("update_config", "msg:UpdateConfigRequest"),
        17: ("bootstrap", "msg:BootstrapRequest"),
        18: ("inject_fault", "msg:InjectFaultRequest"),
        19: ("diagnostic", "msg:DiagnosticRequest"),
        20: ("get_package_status", "msg:GetPackageStatusRequest"),
        21: ("logout", "msg:LogoutRequest"),
        22: ("gossip", "msg:GossipRequest"),
        23: ("trigger_component", "msg:TriggerComponentRequest"),
        24: ("set_plugin_specs", "msg:SetPluginSpecsRequest"),
        25: ("update_token", "msg:UpdateTokenRequest"),
        26: ("get_kap_mtls_status", "msg:GetKAPMTLSStatusRequest"),
        27: ("update_kap_mtls_credentials", "msg:UpdateKAPMTLSCredentialsRequest"),
        28: ("activate_kap_mtls", "msg:ActivateKAPMTLSRequest"),
        29: ("node_credentials", "msg:NodeCredentialsRequest"),
    },
    "DrainNotice": {1: ("reconnect_after_millis", "int64")},
    "Timestamp": {1: ("seconds", "int64"), 2: ("nanos", "int64")},
    "GetHealthStatesRequest": {},
    "GetEventsRequest": {
        1: ("start_time", "msg:Timestamp"),
        2: ("end_time", "msg:Timestamp"),
    },
    "GetMetricsRequest": {1: ("since_nanos", "int64")},
    "UpdateRequest": {1: ("version", "string"), 2: ("since_nanos", "int64")},
    "SetHealthyRequest": {
        1: ("components", "rep_string"),
        2: ("since_nanos", "int64"),
    },
    "RebootRequest": {},
    "UpdateConfigRequest": {1: ("values", "map_str_str")},
    "BootstrapRequest": {
        1: ("timeout_seconds", "int64"),
        2: ("script_base64", "string"),
        3: ("request_present", "bool"),
    },
    "InjectFaultRequest": {
        1: ("request_present", "bool"),
        2: ("xid", "int64"),
        3: ("kernel_message", "msg:KernelMessage"),
    },
    "KernelMessage": {1: ("priority", "string"), 2: ("message", "string")},
    "DiagnosticRequest": {
        1: ("report_id", "string"),
        2: ("type", "string"),
        3: ("timeout_seconds", "int64"),
        4: ("request_present", "bool"),
    },
    "GetPackageStatusRequest": {},
    "LogoutRequest": {},
    "GossipRequest": {},
    "TriggerComponentRequest": {
        1: ("component_name", "string"),
        2: ("tag_name", "string"),
    },
    "SetPluginSpecsRequest": {
        1: ("specs_present", "bool"),
        2: ("specs", "rep_msg:PluginSpec"),
    },
    "PluginSpec": {
        1: ("plugin_name", "string"),
        2: ("plugin_type", "string"),
        3: ("component_list", "rep_string"),
        4: ("component_list_file", "string"),
        5: ("run_mode", "string"),
        6: ("tags", "rep_string"),
        7: ("health_state_plugin", "msg:Plugin"),
        8: ("timeout_nanos", "int64"),
        9: ("interval_nanos", "int64"),
    },
    "Plugin": {
        1: ("steps", "rep_msg:PluginStep"),
        2: ("parser", "msg:PluginOutputParser"),
    },
    "PluginStep": {1: ("name", "string"), 2: ("run_bash_script", "msg:BashScript")},
    "BashScript": {1: ("content_type", "string"), 2: ("script", "string")},
    "PluginOutputParser": {
        1: ("json_paths", "rep_msg:PluginJSONPath"),
        2: ("log_path", "string"),
    },
    "PluginJSONPath": {
        1: ("query", "string"),
        2: ("field", "string"),
        3: ("expect", "msg:PluginMatchRule"),
        4: ("suggested_actions", "map_str_msg:PluginMatchRule"),
    },
    "PluginMatchRule": {1: ("regex", "string")},
    "UpdateTokenRequest": {1: ("token", "string")},
    "GetKAPMTLSStatusRequest": {},
    "UpdateKAPMTLSCredentialsRequest": {
        1: ("certificate_pem", "bytes"),
        2: ("private_key_pem", "bytes"),
        3: ("gateway_ca_pem", "bytes"),
        4: ("gateway_endpoint", "string"),
        5: ("server_name", "string"),
        6: ("client_ca_fingerprint", "string"),
        7: ("gateway_ca_fingerprint", "string"),
    },
    "ActivateKAPMTLSRequest": {},
    "NodeCredentialsRequest": {1: ("kubelet", "msg:KubeletCredentials")},
    "KubeletCredentials": {
        1: ("config", "msg:NodeCredentialFile"),
        2: ("client_certificate", "msg:NodeCredentialFile"),
    },
    "NodeCredentialFile": {
        1: ("path", "string"),
        2: ("contents", "bytes"),
        3: ("mode", "uint32"),
    },
}

_MAP_ENTRY_STR = {1: ("key", "string"), 2: ("value", "string")}


# ---------------------------------------------------------------------------
# generic encode / decode
# ---------------------------------------------------------------------------


def _encode_scalar(field_no: int, kind: str, val: Any) -> bytes:
    if kind in ("uint32", "int64"):
        iv = int(val)
        if iv == 0:
            return b""
        return _tag(field_no, WIRE_VARINT) + encode_varint(iv)
    if kind == "bool":
        if not val:
            return b""
        return _tag(field_no, WIRE_VARINT) + encode_varint(1)
    if kind == "string":
        sv = str(val)
        if not sv:
            return b""
        return _len_field(field_no, sv.encode("utf-8"))
    if kind == "bytes":
        bv = bytes(val)
        if not bv:
            return b""
        return _len_field(field_no, bv)
    raise ValueError(f"unknown scalar kind {kind}")


def encode_message(name: str, obj: Dict[str, Any]) -> bytes:
    desc = MESSAGES[name]
    out = bytearray()
    for field_no in sorted(desc):
        fname, kind = desc[field_no]
        if fname not in obj or obj[fname] is None:
            continue
        val = obj[fname]
        if kind.startswith("msg:"):
            out += _len_field(field_no, encode_message(kind[4:], val))
        elif kind == "rep_string":
            for item in val:
                out += _len_field(field_no, str(item).encode("utf-8"))
        elif kind.startswith("rep_msg:"):
            for item in val:
                out += _len_field(field_no, encode_message(kind[8:], item))
        elif kind == "map_str_str":
            for k, v in val.items():
                entry = _len_field(1, str(k).encode("utf-8")) + _len_field(
                    2, str(v).encode("utf-8")
                )
                out += _len_field(field_no, entry)
        elif kind.startswith("map_str_msg:"):
            sub = kind[len("map_str_msg:"):]
            for k, v in val.items():
                entry = _len_field(1, str(k).encode("utf-8")) + _len_field(
                    2, encode_message(sub, v)
                )
                out += _len_field(field_no, entry)
        else:
            out += _encode_scalar(field_no, kind, val)
    return bytes(out)


def _decode_map_entry(data: bytes, value_msg: str = "") -> Tuple[str, Any]:
    key: str = ""
    value: Any = "" if not value_msg else {}
    pos = 0
    while pos < len(data):
        tag, pos = decode_varint(data, pos)
        field_no, wire = tag >> 3, tag & 7
        if field_no == 1 and wire == WIRE_LEN:
            ln, pos = decode_varint(data, pos)
            key = data[pos : pos + ln].decode("utf-8", "replace")
            pos += ln
        elif field_no == 2 and wire == WIRE_LEN:
            ln, pos = decode_varint(data, pos)
            raw = data[pos : pos + ln]
            pos += ln
            value = decode_message(value_msg, raw) if value_msg else raw.decode(
                "utf-8", "replace"
            )
        else:
            pos = _skip(data, pos, wire)
    return key, value


def decode_message(name: str, data: bytes) -> Dict[str, Any]:
    desc = MESSAGES[name]
    out: Dict[str, Any] = {}
    pos = 0
    while pos < len(data):
        tag, pos = decode_varint(data, pos)
        field_no, wire = tag >> 3, tag & 7
        if field_no not in desc:
            pos = _skip(data, pos, wire)
            continue
        fname, kind = desc[field_no]
        if kind in ("uint32", "int64", "bool"):
            val, pos = decode_varint(data, pos)
            if kind == "bool":
                out[fname] = bool(val)
            elif kind == "int64":
                out[fname] = _to_signed64(val)
            else:
                out[fname] = val
        elif wire == WIRE_LEN:
            ln, pos = decode_varint(data, pos)
            raw = data[pos : pos + ln]
            pos += ln
            if kind == "string":
                out[fname] = raw.decode("utf-8", "replace")
            elif kind == "bytes":
                out[fname] = raw
            elif kind.startswith("msg:"):
                out[fname] = decode_message(kind[4:], raw)
            elif kind == "rep_string":
                out.setdefault(fname, []).append(raw.decode("utf-8", "replace"))
            elif kind.startswith("rep_msg:"):
                out.setdefault(fname, []).append(decode_message(kind[8:], raw))
            elif kind == "map_str_str":
                k, v = _decode_map_entry(raw)
                out.setdefault(fname, {})[k] = v
            elif kind.startswith("map_str_msg:"):
                k, v = _decode_map_entry(raw, kind[len("map_str_msg:"):])
                out.setdefault(fname, {})[k] = v
            else:
                pass  # schema/wire mismatch: ignore
        else:
            pos = _skip(data, pos, wire)
    return out


# ---------------------------------------------------------------------------
# adapter: ManagerPacket -> v1 dispatch frame; AgentPacket builders
# ---------------------------------------------------------------------------

PROTOCOL_REVISION = 1

_PRIORITY_NAMES = {
    "KERN_EMERG": 0, "KERN_ALERT": 1, "KERN_CRIT": 2, "KERN_ERR": 3,
    "KERN_WARNING": 4, "KERN_NOTICE": 5, "KERN_INFO": 6, "KERN_DEBUG": 7,
    "emerg": 0, "alert": 1, "crit": 2, "err": 3, "warning": 4, "notice": 5,
    "info": 6, "debug": 7,
}


def _ts_to_rfc3339(ts: Optional[Dict[str, int]]) -> str:
    if not ts or not ts.get("seconds"):
        return ""
    dt = datetime.datetime.fromtimestamp(
        ts["seconds"] + ts.get("nanos", 0) / 1e9, tz=datetime.timezone.utc
    )
    return dt.strftime("%Y-%m-%dT%H:%M:%S.%f") + "Z"


def _plugin_spec_to_dict(spec: Dict[str, Any]) -> Dict[str, Any]:
    """PluginSpec proto -> the YAML-shaped dict Spec.from_dict accepts."""
    hsp = spec.get("health_state_plugin") or {}
    steps = [
        {
            "name": st.get("name", ""),
            "run_bash_script": {
                "content_type": (st.get("run_bash_script") or {}).get(
                    "content_type", "plaintext"
                ),
                "script": (st.get("run_bash_script") or {}).get("script", ""),
            },
        }
        for st in hsp.get("steps", [])
    ]
    parser = hsp.get("parser") or {}
    json_paths = [
        {
            "query": r.get("query", ""),
            "field": r.get("field", ""),
            **(
                {"expect": {"regex": r["expect"].get("regex", "")}}
                if r.get("expect")
                else {}
            ),
            **(
                {
                    "suggested_actions": {
                        k: {"regex": v.get("regex", "")}
                        for k, v in r["suggested_actions"].items()
                    }
                }
                if r.get("suggested_actions")
                else {}
            ),
        }
        for r in parser.get("json_paths", [])
    ]
    out: Dict[str, Any] = {
        "plugin_name": spec.get("plugin_name", ""),
        "plugin_type": spec.get("plugin_type", "component"),
        "run_mode": spec.get("run_mode", "auto"),
        "tags": spec.get("tags", []),
        "component_list": spec.get("component_list", []),
        "component_list_file": spec.get("component_list_file", ""),
        "health_state_plugin": {
            "steps": steps,
            "parser": {
                "json_paths": json_paths,
                "log_path": parser.get("log_path", ""),
            },
        },
    }
    if spec.get("timeout_nanos"):
        out["timeout"] = f"{spec['timeout_nanos'] / 1e9:g}s"
    if spec.get("interval_nanos"):
        out["interval"] = f"{spec['interval_nanos'] / 1e9:g}s"
    return out


# manager oneof field -> (v1 method, payload transform)
def manager_packet_to_frame(data: bytes) -> Optional[Dict[str, Any]]:
    """Decode one ManagerPacket; returns a v1 dispatch frame
    {"req_id", "method", "data"}, a control dict {"_control": ...} for
    hello_ack/drain_notice, or None for an empty packet."""
    pkt = decode_message("ManagerPacket", data)
    req_id = pkt.get("request_id", "")
    if "hello_ack" in pkt:
        return {"_control": "hello_ack", **pkt["hello_ack"]}
    if "drain_notice" in pkt:
        return {"_control": "drain_notice", **pkt["drain_notice"]}

    def frame(method: str, payload: Dict[str, Any]) -> Dict[str, Any]:
        return {"req_id": req_id, "method": method, "data": payload}

    if "get_health_states" in pkt:
        return frame("states", {})
    if "get_events" in pkt:
        p = pkt["get_events"]
        payload = {}
        start = _ts_to_rfc3339(p.get("start_time"))
        if start:
            payload["startTime"] = start
        return frame("events", payload)
    if "get_metrics" in pkt:
        n = pkt["get_metrics"].get("since_nanos", 0)
        payload = {}
        if n:
            payload["since"] = _ts_to_rfc3339({"seconds": n // 10**9})
        return frame("metrics", payload)
    if "update" in pkt:
        return frame("update", {"version": pkt["update"].get("version", "")})
    if "set_healthy" in pkt:
        comps = pkt["set_healthy"].get("components", [])
        return frame("setHealthy", {"components": comps} if comps else {})
    if "reboot" in pkt:
        return frame("reboot", {})
    if "update_config" in pkt:
        return frame("updateConfig", pkt["update_config"].get("values", {}))
    if "bootstrap" in pkt:
        p = pkt["bootstrap"]
        return frame(
            "bootstrap",
            {
                "script": p.get("script_base64", ""),
                "timeout_seconds": p.get("timeout_seconds", 120) or 120,
            },
        )
    if "inject_fault" in pkt:
        p = pkt["inject_fault"]
        km = p.get("kernel_message")
        payload: Dict[str, Any] = {}
        if km:
            pr = km.get("priority", "")
            payload["kernel_message"] = {
                "message": km.get("message", ""),
                "priority": _PRIORITY_NAMES.get(pr, int(pr) if str(pr).isdigit() else 2),
            }
        return frame("injectFault", payload)
    if "diagnostic" in pkt:
        p = pkt["diagnostic"]
        return frame(
            "diagnostic",
            {
                "report_id": p.get("report_id", ""),
                "type": p.get("type", "amd_bug_report"),
                "timeout_seconds": p.get("timeout_seconds", 600) or 600,
            },
        )
    if "get_package_status" in pkt:
        return frame("packageStatus", {})
    if "logout" in pkt:
        return frame("logout", {})
    if "gossip" in pkt:
        return frame("gossip", {})
    if "trigger_component" in pkt:
        p = pkt["trigger_component"]
        return frame(
            "triggerComponent",
            {
                "component_name": p.get("component_name", ""),
                "tag_name": p.get("tag_name", ""),
            },
        )
    if "set_plugin_specs" in pkt:
        specs = [
            _plugin_spec_to_dict(s)
            for s in pkt["set_plugin_specs"].get("specs", [])
        ]
        return frame("setPluginSpecs", {"specs": specs})
    if "update_token" in pkt:
        return frame("updateToken", {"token": pkt["update_token"].get("token", "")})
    if "get_kap_mtls_status" in pkt:
        return frame("kapMTLSStatus", {})
    if "update_kap_mtls_credentials" in pkt:
        import base64 as _b64

        p = pkt["update_kap_mtls_credentials"]
        return frame(
            "updateKAPMTLSCredentials",
            {
                "cert": _b64.b64encode(p.get("certificate_pem", b"")).decode(),
                "key": _b64.b64encode(p.get("private_key_pem", b"")).decode(),
                "gateway_ca": _b64.b64encode(p.get("gateway_ca_pem", b"")).decode(),
                "gateway_endpoint": p.get("gateway_endpoint", ""),
                "server_name": p.get("server_name", ""),
            },
        )
    if "activate_kap_mtls" in pkt:
        return frame("activateKAPMTLS", {})
    if "node_credentials" in pkt:
        kub = pkt["node_credentials"].get("kubelet", {})

        def _file(f: Optional[Dict[str, Any]]) -> Optional[Dict[str, Any]]:
            if not f:
                return None
            import base64 as _b64

            return {
                "path": f.get("path", ""),
                "contents": _b64.b64encode(f.get("contents", b"")).decode(),
                "mode": f.get("mode", 0),
            }

        return frame(
            "nodeCredentials",
            {
                "kubelet": {
                    "config": _file(kub.get("config")),
                    "client_certificate": _file(kub.get("client_certificate")),
                }
            },
        )
    return None


def hello_bytes(agent_version: str, capabilities: Optional[List[str]] = None) -> bytes:
    return encode_message(
        "AgentPacket",
        {
            "hello": {
                "min_protocol_revision": PROTOCOL_REVISION,
                "max_protocol_revision": PROTOCOL_REVISION,
                "agent_version": agent_version,
                "capabilities": capabilities or [],
            }
        },
    )


def result_bytes(request_id: str, data: Any) -> bytes:
    return encode_message(
        "AgentPacket",
        {
            "result": {
                "request_id": request_id,
                "payload_json": json.dumps(data, default=str).encode("utf-8"),
            }
        },
    )
