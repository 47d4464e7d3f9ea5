"""Cross-validate the hand-written session-v2 codec against the real
google.protobuf runtime (available in the image; protoc is not).

The schema is rebuilt at test time as a FileDescriptorProto and message
classes are created via message_factory — so every byte the hand codec
produces is parsed by Google's implementation and vice versa. Oneof
declarations are omitted (wire format is identical for plain singular
fields) and map fields are declared as repeated key/value entry messages
(again wire-identical), which is exactly what the codec emits.
"""

import json

import pytest

pb = pytest.importorskip("google.protobuf")

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory  # noqa: E402

from gpud_amd.session import protowire as pw  # noqa: E402

T = descriptor_pb2.FieldDescriptorProto


def _msg(fd, name):
    m = fd.message_type.add()
    m.name = name
    return m


def _field(m, num, name, ftype, type_name="", label=T.LABEL_OPTIONAL):
    f = m.field.add()
    f.name = name
    f.number = num
    f.type = ftype
    f.label = label
    if type_name:
        f.type_name = type_name
    return f


@pytest.fixture(scope="module")
def classes():
    fd = descriptor_pb2.FileDescriptorProto()
    fd.name = "session_v2_crosscheck.proto"
    fd.package = "gpud.session.v2.crosscheck"
    P = ".gpud.session.v2.crosscheck."

    m = _msg(fd, "Hello")
    _field(m, 1, "min_protocol_revision", T.TYPE_UINT32)
    _field(m, 2, "max_protocol_revision", T.TYPE_UINT32)
    _field(m, 3, "agent_version", T.TYPE_STRING)
    _field(m, 4, "max_receive_message_bytes", T.TYPE_UINT32)
    _field(m, 5, "capabilities", T.TYPE_STRING, label=T.LABEL_REPEATED)

    m = _msg(fd, "HelloAck")
    _field(m, 1, "protocol_revision", T.TYPE_UINT32)
    _field(m, 2, "manager_instance_id", T.TYPE_STRING)
    _field(m, 3, "max_receive_message_bytes", T.TYPE_UINT32)

    m = _msg(fd, "Result")
    _field(m, 1, "request_id", T.TYPE_STRING)
    _field(m, 2, "payload_json", T.TYPE_BYTES)

    m = _msg(fd, "AgentPacket")
    _field(m, 1, "hello", T.TYPE_MESSAGE, P + "Hello")
    _field(m, 2, "result", T.TYPE_MESSAGE, P + "Result")

    m = _msg(fd, "Timestamp")
    _field(m, 1, "seconds", T.TYPE_INT64)
    _field(m, 2, "nanos", T.TYPE_INT64)

    m = _msg(fd, "GetEventsRequest")
    _field(m, 1, "start_time", T.TYPE_MESSAGE, P + "Timestamp")
    _field(m, 2, "end_time", T.TYPE_MESSAGE, P + "Timestamp")

    m = _msg(fd, "ValuesEntry")
    _field(m, 1, "key", T.TYPE_STRING)
    _field(m, 2, "value", T.TYPE_STRING)

    m = _msg(fd, "UpdateConfigRequest")
    _field(m, 1, "values", T.TYPE_MESSAGE, P + "ValuesEntry",
           label=T.LABEL_REPEATED)

    m = _msg(fd, "SetHealthyRequest")
    _field(m, 1, "components", T.TYPE_STRING, label=T.LABEL_REPEATED)
    _field(m, 2, "since_nanos", T.TYPE_INT64)

    m = _msg(fd, "DiagnosticRequest")
    _field(m, 1, "report_id", T.TYPE_STRING)
    _field(m, 2, "type", T.TYPE_STRING)
    _field(m, 3, "timeout_seconds", T.TYPE_INT64)
    _field(m, 4, "request_present", T.TYPE_BOOL)

    m = _msg(fd, "KernelMessage")
    _field(m, 1, "priority", T.TYPE_STRING)
    _field(m, 2, "message", T.TYPE_STRING)

    m = _msg(fd, "InjectFaultRequest")
    _field(m, 1, "request_present", T.TYPE_BOOL)
    _field(m, 2, "xid", T.TYPE_INT64)
    _field(m, 3, "kernel_message", T.TYPE_MESSAGE, P + "KernelMessage")

    m = _msg(fd, "TriggerComponentRequest")
    _field(m, 1, "component_name", T.TYPE_STRING)
    _field(m, 2, "tag_name", T.TYPE_STRING)

    m = _msg(fd, "BashScript")
    _field(m, 1, "content_type", T.TYPE_STRING)
    _field(m, 2, "script", T.TYPE_STRING)

    m = _msg(fd, "PluginStep")
    _field(m, 1, "name", T.TYPE_STRING)
    _field(m, 2, "run_bash_script", T.TYPE_MESSAGE, P + "BashScript")

    m = _msg(fd, "PluginMatchRule")
    _field(m, 1, "regex", T.TYPE_STRING)

    m = _msg(fd, "ActionsEntry")
    _field(m, 1, "key", T.TYPE_STRING)
    _field(m, 2, "value", T.TYPE_MESSAGE, P + "PluginMatchRule")

    m = _msg(fd, "PluginJSONPath")
    _field(m, 1, "query", T.TYPE_STRING)
    _field(m, 2, "field", T.TYPE_STRING)
    _field(m, 3, "expect", T.TYPE_MESSAGE, P + "PluginMatchRule")
    _field(m, 4, "suggested_actions", T.TYPE_MESSAGE, P + "ActionsEntry",
           label=T.LABEL_REPEATED)

    m = _msg(fd, "PluginOutputParser")
    _field(m, 1, "json_paths", T.TYPE_MESSAGE, P + "PluginJSONPath",
           label=T.LABEL_REPEATED)
    _field(m, 2, "log_path", T.TYPE_STRING)

    m = _msg(fd, "Plugin")
    _field(m, 1, "steps", T.TYPE_MESSAGE, P + "PluginStep",
           label=T.LABEL_REPEATED)
    _field(m, 2, "parser", T.TYPE_MESSAGE, P + "PluginOutputParser")

    m = _msg(fd, "PluginSpec")
    _field(m, 1, "plugin_name", T.TYPE_STRING)
    _field(m, 2, "plugin_type", T.TYPE_STRING)
    _field(m, 3, "component_list", T.TYPE_STRING, label=T.LABEL_REPEATED)
    _field(m, 4, "component_list_file", T.TYPE_STRING)
    _field(m, 5, "run_mode", T.TYPE_STRING)
    _field(m, 6, "tags", T.TYPE_STRING, label=T.LABEL_REPEATED)
    _field(m, 7, "health_state_plugin", T.TYPE_MESSAGE, P + "Plugin")
    _field(m, 8, "timeout_nanos", T.TYPE_INT64)
    _field(m, 9, "interval_nanos", T.TYPE_INT64)

    m = _msg(fd, "DrainNotice")
    _field(m, 1, "reconnect_after_millis", T.TYPE_INT64)

    m = _msg(fd, "GetHealthStatesRequest")

    m = _msg(fd, "SetPluginSpecsRequest")
    _field(m, 1, "specs_present", T.TYPE_BOOL)
    _field(m, 2, "specs", T.TYPE_MESSAGE, P + "PluginSpec",
           label=T.LABEL_REPEATED)

    m = _msg(fd, "ManagerPacket")
    _field(m, 1, "hello_ack", T.TYPE_MESSAGE, P + "HelloAck")
    _field(m, 3, "drain_notice", T.TYPE_MESSAGE, P + "DrainNotice")
    _field(m, 4, "request_id", T.TYPE_STRING)
    _field(m, 10, "get_health_states", T.TYPE_MESSAGE,
           P + "GetHealthStatesRequest")
    _field(m, 11, "get_events", T.TYPE_MESSAGE, P + "GetEventsRequest")
    _field(m, 16, "update_config", T.TYPE_MESSAGE, P + "UpdateConfigRequest")
    _field(m, 18, "inject_fault", T.TYPE_MESSAGE, P + "InjectFaultRequest")
    _field(m, 19, "diagnostic", T.TYPE_MESSAGE, P + "DiagnosticRequest")
    _field(m, 23, "trigger_component", T.TYPE_MESSAGE,
           P + "TriggerComponentRequest")
    _field(m, 24, "set_plugin_specs", T.TYPE_MESSAGE,
           P + "SetPluginSpecsRequest")

    pool = descriptor_pool.DescriptorPool()
    pool.Add(fd)
    out = {}
    for name in [x.name for x in fd.message_type]:
        desc = pool.FindMessageTypeByName(f"gpud.session.v2.crosscheck.{name}")
        out[name] = message_factory.GetMessageClass(desc)
    return out


def _fill(msg, d):
    """Recursively fill a protobuf message from the codec's dict shape."""
    for fname, val in d.items():
        fdesc = msg.DESCRIPTOR.fields_by_name[fname]
        if fdesc.label == fdesc.LABEL_REPEATED:
            if fdesc.message_type is not None:
                if isinstance(val, dict):  # map declared as entry list
                    for k, v in val.items():
                        entry = getattr(msg, fname).add()
                        entry.key = k
                        if isinstance(v, dict):
                            _fill(entry.value, v)
                        else:
                            entry.value = v
                else:
                    for item in val:
                        _fill(getattr(msg, fname).add(), item)
            else:
                getattr(msg, fname).extend(val)
        elif fdesc.message_type is not None:
            _fill(getattr(msg, fname), val)
        elif fdesc.type == fdesc.TYPE_BYTES:
            setattr(msg, fname, bytes(val))
        else:
            setattr(msg, fname, val)


SAMPLES = [
    ("Hello", {"min_protocol_revision": 1, "max_protocol_revision": 1,
               "agent_version": "0.1.0", "capabilities": ["a", "b"]}),
    ("Result", {"request_id": "r-42",
                "payload_json": json.dumps({"ok": 1}).encode()}),
    ("AgentPacket", {"result": {"request_id": "x", "payload_json": b"{}"}}),
    ("GetEventsRequest", {"start_time": {"seconds": 1_757_000_000,
                                         "nanos": 500}}),
    ("UpdateConfigRequest", {"values": {"k1": "v1", "k2": "v2"}}),
    ("SetHealthyRequest", {"components": ["cpu", "memory"],
                           "since_nanos": 12345}),
    ("DiagnosticRequest", {"report_id": "rep", "type": "amd_bug_report",
                           "timeout_seconds": 600, "request_present": True}),
    ("InjectFaultRequest", {"request_present": True,
                            "kernel_message": {"priority": "KERN_ERR",
                                               "message": "boom"}}),
    ("TriggerComponentRequest", {"component_name": "cpu", "tag_name": ""}),
    ("DrainNotice", {"reconnect_after_millis": 2500}),
    ("PluginSpec", {
        "plugin_name": "p", "plugin_type": "component", "run_mode": "auto",
        "component_list": ["a:p1", "b#manual"],
        "tags": ["t"],
        "health_state_plugin": {
            "steps": [{"name": "s", "run_bash_script": {
                "content_type": "plaintext", "script": "echo"}}],
            "parser": {
                "json_paths": [{"query": "q", "field": "f",
                                "expect": {"regex": "^x$"},
                                "suggested_actions": {
                                    "REBOOT_SYSTEM": {"regex": ".*"}}}],
                "log_path": "/tmp/l",
            },
        },
        "timeout_nanos": 5_000_000_000,
    }),
    ("ManagerPacket", {"request_id": "q9",
                       "update_config": {"values": {"a": "1"}}}),
    ("ManagerPacket", {"hello_ack": {"protocol_revision": 1,
                                     "manager_instance_id": "mgr"}}),
]


@pytest.mark.parametrize("name,sample", SAMPLES,
                         ids=[f"{n}-{i}" for i, (n, _) in enumerate(SAMPLES)])
def test_hand_encode_google_decode(classes, name, sample):
    """Bytes from the hand codec parse losslessly in Google's runtime."""
    mine = pw.encode_message(name, sample)
    theirs = classes[name]()
    theirs.ParseFromString(mine)
    # re-serialize with Google's runtime and decode with the hand codec
    back = pw.decode_message(name, theirs.SerializeToString())

    def norm(d):
        if isinstance(d, dict):
            return {k: norm(v) for k, v in d.items()}
        if isinstance(d, list):
            return [norm(x) for x in d]
        return d

    # drop proto3 defaults the samples carried explicitly (e.g. "" strings)
    def strip_defaults(d):
        if isinstance(d, dict):
            return {
                k: strip_defaults(v)
                for k, v in d.items()
                if v not in ("", 0, False, [], {}) or isinstance(v, dict)
            }
        if isinstance(d, list):
            return [strip_defaults(x) for x in d]
        return d

    assert norm(back) == strip_defaults(norm(sample))


def test_google_encode_hand_decode_manager_flow(classes):
    """A ManagerPacket built and serialized by Google's runtime drives the
    hand adapter end-to-end."""
    pkt = classes["ManagerPacket"]()
    pkt.request_id = "g1"
    pkt.diagnostic.report_id = "rep-7"
    pkt.diagnostic.type = "amd_bug_report"
    pkt.diagnostic.timeout_seconds = 60
    frame = pw.manager_packet_to_frame(pkt.SerializeToString())
    assert frame == {
        "req_id": "g1",
        "method": "diagnostic",
        "data": {"report_id": "rep-7", "type": "amd_bug_report",
                 "timeout_seconds": 60},
    }

    hello = pw.decode_message(
        "AgentPacket", pw.hello_bytes("9.9.9")
    )
    ap = classes["AgentPacket"]()
    ap.ParseFromString(pw.hello_bytes("9.9.9"))
    assert ap.hello.agent_version == "9.9.9" == hello["hello"]["agent_version"]
