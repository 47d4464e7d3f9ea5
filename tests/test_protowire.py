"""Protobuf wire-format tests for the session v2 codec.

Golden byte vectors are hand-computed from the proto3 wire spec so the
codec is proven against the FORMAT, not merely self-consistent
(reference schema: pkg/session/v2/session.proto)."""

import json

import pytest

from gpud_amd.session import protowire as pw


def test_varint_roundtrip():
    for n in (0, 1, 127, 128, 300, 2**31, 2**63 - 1):
        b = pw.encode_varint(n)
        got, pos = pw.decode_varint(b, 0)
        assert got == n and pos == len(b)
    # negative int64 is 10 bytes two's complement
    b = pw.encode_varint(-1)
    assert len(b) == 10
    got, _ = pw.decode_varint(b, 0)
    assert pw._to_signed64(got) == -1


def test_golden_hello():
    # Hello{min=1 max=2 agent_version="x"}:
    #   field1 varint: tag 0x08 val 0x01
    #   field2 varint: tag 0x10 val 0x02
    #   field3 LEN:    tag 0x1a len 1 'x'
    b = pw.encode_message(
        "Hello",
        {"min_protocol_revision": 1, "max_protocol_revision": 2,
         "agent_version": "x"},
    )
    assert b == b"\x08\x01\x10\x02\x1a\x01x"


def test_golden_manager_packet():
    # ManagerPacket{request_id="r1" get_health_states{}}:
    #   field4 LEN: tag 0x22 len 2 "r1"; field10 LEN: tag 0x52 len 0
    b = pw.encode_message(
        "ManagerPacket", {"request_id": "r1", "get_health_states": {}}
    )
    assert b == b"\x22\x02r1\x52\x00"
    # decoding accepts any field order
    back = pw.decode_message("ManagerPacket", b"\x52\x00\x22\x02r1")
    assert back == {"request_id": "r1", "get_health_states": {}}


def test_golden_map_field():
    # UpdateConfigRequest{values:{"a":"b"}}: field1 LEN of entry
    #   entry = field1 LEN "a" + field2 LEN "b" = 0a 01 61 12 01 62
    b = pw.encode_message("UpdateConfigRequest", {"values": {"a": "b"}})
    assert b == b"\x0a\x06\x0a\x01a\x12\x01b"
    assert pw.decode_message("UpdateConfigRequest", b) == {"values": {"a": "b"}}


def test_golden_result_bytes():
    b = pw.result_bytes("id7", {"ok": True})
    # AgentPacket.result is field 2: tag 0x12
    assert b[0] == 0x12
    back = pw.decode_message("AgentPacket", b)
    assert back["result"]["request_id"] == "id7"
    assert json.loads(back["result"]["payload_json"]) == {"ok": True}


def test_unknown_fields_skipped():
    # a future field 99 (varint) must be skipped, not fatal
    extra = pw._tag(99, 0) + pw.encode_varint(7)
    b = extra + pw.encode_message("Hello", {"agent_version": "v"})
    assert pw.decode_message("Hello", b)["agent_version"] == "v"


def test_nested_plugin_spec_roundtrip():
    spec = {
        "plugin_name": "p1",
        "plugin_type": "component_list",
        "component_list": ["a", "b#manual:/x"],
        "run_mode": "auto",
        "tags": ["t1"],
        "health_state_plugin": {
            "steps": [
                {"name": "s1",
                 "run_bash_script": {"content_type": "plaintext",
                                     "script": "echo hi"}}
            ],
            "parser": {
                "json_paths": [
                    {"query": "r", "field": "r",
                     "expect": {"regex": "^ok$"},
                     "suggested_actions": {
                         "REBOOT_SYSTEM": {"regex": ".*reboot.*"}}}
                ],
                "log_path": "/var/log/${PLUGIN}.log",
            },
        },
        "timeout_nanos": 30_000_000_000,
        "interval_nanos": 600_000_000_000,
    }
    b = pw.encode_message("PluginSpec", spec)
    back = pw.decode_message("PluginSpec", b)
    assert back == spec
    # and the adapter turns it into a Spec.from_dict-compatible dict
    d = pw._plugin_spec_to_dict(back)
    from gpud_amd.pkg.custom_plugins import Spec

    s = Spec.from_dict(d)
    assert s.validate() is None
    assert s.timeout_seconds == 30.0
    assert s.interval_seconds == 600.0
    assert s.log_path == "/var/log/${PLUGIN}.log"
    assert s.json_paths[0].expect_regex == "^ok$"


@pytest.mark.parametrize(
    "oneof,inner,method",
    [
        ("get_health_states", {}, "states"),
        ("reboot", {}, "reboot"),
        ("gossip", {}, "gossip"),
        ("logout", {}, "logout"),
        ("get_package_status", {}, "packageStatus"),
        ("get_kap_mtls_status", {}, "kapMTLSStatus"),
        ("activate_kap_mtls", {}, "activateKAPMTLS"),
        ("update_token", {"token": "t"}, "updateToken"),
        ("trigger_component", {"component_name": "cpu"}, "triggerComponent"),
        ("update_config", {"values": {"poll_interval_seconds": "30"}},
         "updateConfig"),
        ("set_healthy", {"components": ["cpu"]}, "setHealthy"),
        ("diagnostic", {"report_id": "r9", "type": "amd_bug_report",
                        "timeout_seconds": 60}, "diagnostic"),
    ],
)
def test_manager_packet_to_frame_methods(oneof, inner, method):
    b = pw.encode_message("ManagerPacket", {"request_id": "q", oneof: inner})
    frame = pw.manager_packet_to_frame(b)
    assert frame["method"] == method, frame
    assert frame["req_id"] == "q"


def test_manager_control_packets():
    b = pw.encode_message(
        "ManagerPacket",
        {"hello_ack": {"protocol_revision": 1, "manager_instance_id": "m1"}},
    )
    ctl = pw.manager_packet_to_frame(b)
    assert ctl["_control"] == "hello_ack" and ctl["manager_instance_id"] == "m1"
    b = pw.encode_message(
        "ManagerPacket", {"drain_notice": {"reconnect_after_millis": 1500}}
    )
    ctl = pw.manager_packet_to_frame(b)
    assert ctl["_control"] == "drain_notice"
    assert ctl["reconnect_after_millis"] == 1500


def test_inject_fault_priority_mapping():
    b = pw.encode_message(
        "ManagerPacket",
        {"request_id": "i", "inject_fault": {
            "request_present": True,
            "kernel_message": {"priority": "KERN_ERR", "message": "boom"}}},
    )
    frame = pw.manager_packet_to_frame(b)
    assert frame["method"] == "injectFault"
    assert frame["data"]["kernel_message"] == {"message": "boom", "priority": 3}


def test_get_events_timestamp_conversion():
    b = pw.encode_message(
        "ManagerPacket",
        {"request_id": "e",
         "get_events": {"start_time": {"seconds": 1_757_000_000}}},
    )
    frame = pw.manager_packet_to_frame(b)
    assert frame["method"] == "events"
    assert frame["data"]["startTime"].startswith("2025-09-04T")


def test_protowire_fuzz_roundtrip():
    """Property fuzz: random dicts for every message round-trip through
    encode/decode losslessly (hypothesis-driven)."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    scalar = {
        "uint32": st.integers(min_value=0, max_value=2**32 - 1),
        "int64": st.integers(min_value=-(2**63), max_value=2**63 - 1),
        "bool": st.booleans(),
        "string": st.text(max_size=20),
        "bytes": st.binary(max_size=20),
    }

    def msg_strategy(name, depth=0):
        desc = pw.MESSAGES[name]
        fields = {}
        for _no, (fname, kind) in desc.items():
            if kind in scalar:
                fields[fname] = scalar[kind]
            elif kind == "rep_string":
                fields[fname] = st.lists(st.text(max_size=8), max_size=3)
            elif kind.startswith("msg:") and depth < 2:
                fields[fname] = msg_strategy(kind[4:], depth + 1)
            elif kind.startswith("rep_msg:") and depth < 2:
                fields[fname] = st.lists(
                    msg_strategy(kind[8:], depth + 1), max_size=2
                )
            elif kind == "map_str_str":
                fields[fname] = st.dictionaries(
                    st.text(min_size=1, max_size=6), st.text(max_size=6),
                    max_size=3,
                )
        return st.fixed_dictionaries({}, optional=fields)

    def strip_empty(name, d):
        """proto3 has no presence for empty scalars/maps/lists: drop values
        that encode to nothing. Empty MESSAGE fields keep presence (a zero-
        length LEN field is still emitted)."""
        desc = {fn: kind for _no, (fn, kind) in pw.MESSAGES[name].items()}
        out = {}
        for k, v in d.items():
            kind = desc[k]
            if kind.startswith("msg:"):
                out[k] = strip_empty(kind[4:], v)  # presence kept
            elif kind.startswith("rep_msg:"):
                v2 = [strip_empty(kind[8:], x) for x in v]
                if v2:
                    out[k] = v2
            elif kind.startswith("map_") or kind == "rep_string":
                if v:
                    out[k] = v
            else:
                if v not in ("", 0, False, b""):
                    out[k] = v
        return out

    for name in ("Hello", "Result", "UpdateConfigRequest", "PluginSpec",
                 "ManagerPacket", "DiagnosticRequest"):
        def make_check(_name):
            @settings(max_examples=40, deadline=None)
            @given(sample=msg_strategy(_name))
            def check(sample):
                b = pw.encode_message(_name, sample)
                back = pw.decode_message(_name, b)
                assert back == strip_empty(_name, sample), (_name, sample, back)

            return check

        make_check(name)()
