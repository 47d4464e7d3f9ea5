"""Property-based tests (hypothesis) for the parsing/crypto kernels of the
daemon — the components most exposed to adversarial/arbitrary input:
kmsg records, duration strings, JSON dot-paths, ed25519, event ordering."""

import datetime
import string

from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from gpud_amd.apiv1.types import Event, utcnow
from gpud_amd.pkg.kmsg.parser import parse_line


@settings(max_examples=200, deadline=None)
@given(
    priority=st.integers(min_value=0, max_value=191),
    seq=st.integers(min_value=0, max_value=2**40),
    ts=st.integers(min_value=0, max_value=2**50),
    msg=st.text(
        alphabet=string.printable.replace("\n", "").replace("\r", ""),
        max_size=200,
    ),
)
def test_kmsg_parse_roundtrip(priority, seq, ts, msg):
    line = f"{priority},{seq},{ts},-;{msg}"
    m = parse_line(line, boot_time_epoch=0.0)
    assert m is not None
    assert m.priority == priority
    assert m.sequence == seq
    assert m.timestamp_us == ts
    assert m.message == msg
    assert 0 <= m.severity <= 7
    assert m.facility == priority >> 3


@settings(max_examples=100, deadline=None)
@given(st.text(max_size=80))
def test_kmsg_parse_never_raises(garbage):
    # arbitrary input may return None but must never raise
    parse_line(garbage, boot_time_epoch=0.0)


@settings(max_examples=100, deadline=None)
@given(
    val=st.floats(min_value=0, max_value=10_000, allow_nan=False),
    unit=st.sampled_from(["ms", "s", "m", "h", ""]),
)
def test_parse_duration_units(val, unit):
    from gpud_amd.pkg.custom_plugins import parse_duration

    mult = {"ms": 0.001, "s": 1, "m": 60, "h": 3600, "": 1}[unit]
    got = parse_duration(f"{val}{unit}")
    assert abs(got - val * mult) < 1e-6 * max(1.0, val * mult)


@settings(max_examples=60, deadline=None)
@given(st.text(max_size=40))
def test_parse_duration_never_raises(s):
    from gpud_amd.pkg.custom_plugins import parse_duration

    assert parse_duration(s) >= 0


@settings(max_examples=50, deadline=None)
@given(
    payload=st.dictionaries(
        st.text(alphabet=string.ascii_letters, min_size=1, max_size=8),
        st.one_of(st.integers(), st.text(max_size=10)),
        max_size=4,
    ),
    path=st.text(alphabet=string.ascii_letters + ".", max_size=20),
)
def test_dotpath_dig_never_raises(payload, path):
    from gpud_amd.pkg.custom_plugins import _dig

    _dig(payload, path)  # may be None, must not raise


@settings(max_examples=20, deadline=None)
@given(
    message=st.binary(min_size=0, max_size=256),
    seed=st.binary(min_size=32, max_size=32),
)
def test_ed25519_sign_verify_property(message, seed):
    from gpud_amd.pkg import distsign

    _, pub = distsign.generate_keypair(seed)
    sig = distsign.sign(message, seed)
    assert distsign.verify(message, sig, pub)
    # a flipped message bit must not verify
    if message:
        tampered = bytes([message[0] ^ 1]) + message[1:]
        assert not distsign.verify(tampered, sig, pub)


@settings(
    max_examples=25,
    deadline=None,
    suppress_health_check=[HealthCheck.function_scoped_fixture],
)
@given(
    offsets=st.lists(
        st.integers(min_value=0, max_value=86_400), min_size=1, max_size=30
    )
)
def test_eventstore_returns_desc_order(mem_db, offsets):
    from gpud_amd.pkg.eventstore import Store

    rw, ro = mem_db
    store = Store(rw, ro)
    bucket = store.bucket("prop-order", disable_purge=True)
    base = utcnow() - datetime.timedelta(days=2)
    for off in offsets:
        bucket.insert(
            Event(
                time=base + datetime.timedelta(seconds=off),
                name="e",
                type="Info",
                message=str(off),
            )
        )
    evs = bucket.get(base - datetime.timedelta(seconds=1))
    times = [e.time for e in evs]
    assert times == sorted(times, reverse=True)
    assert len(evs) == len(offsets)
    # cleanup for the next example (function-scoped via new table? same
    # table accumulates across examples — purge everything)
    bucket.purge(int((base + datetime.timedelta(days=5)).timestamp()))
    store.close()


@settings(max_examples=60, deadline=None)
@given(
    doc=st.recursive(
        st.one_of(st.none(), st.booleans(), st.integers(), st.text(max_size=12)),
        lambda children: st.one_of(
            st.lists(children, max_size=4),
            st.dictionaries(st.text(max_size=8), children, max_size=4),
        ),
        max_leaves=12,
    )
)
def test_plugin_spec_loader_never_crashes(tmp_path_factory, doc):
    """Arbitrary YAML documents either load or raise ValueError/TypeError —
    never an unhandled crash (the specs file is operator-supplied)."""
    import yaml as _yaml

    from gpud_amd.pkg.custom_plugins import load_specs

    p = tmp_path_factory.mktemp("specs") / "s.yaml"
    p.write_text(_yaml.safe_dump(doc))
    try:
        specs = load_specs(str(p))
        assert isinstance(specs, list)
    except ValueError:
        pass  # clean rejection is fine; anything else propagates and fails


@settings(max_examples=300, deadline=None)
@given(line=st.text(max_size=400))
def test_ras_catalog_match_total(line):
    """The 99-signature catalog must never raise on arbitrary text and
    must return a catalog Detail when it matches."""
    from gpud_amd.pkg.ras_catalog import CATALOG, match

    res = match(line)
    if res is not None:
        detail, groups = res
        assert any(d.name == detail.name for d in CATALOG)
        assert isinstance(groups, dict)


@settings(max_examples=150, deadline=None)
@given(text=st.text(max_size=600))
def test_component_kmsg_matchers_total(text):
    """disk/nfs/memory component matchers never raise on arbitrary
    input."""
    from gpud_amd.components.host.disk import match_disk_kmsg
    from gpud_amd.components.host.memory import match_memory_kmsg
    from gpud_amd.components.host.nfs import match_nfs_kmsg

    for fn in (match_disk_kmsg, match_nfs_kmsg, match_memory_kmsg):
        res = fn(text)
        assert res is None or res.name


@settings(max_examples=100, deadline=None)
@given(
    rows=st.lists(
        st.tuples(
            st.sampled_from(
                ["bpf_jit_alloc_exec+0xe/0x20", "vmap", "irq_init", ""]),
            st.integers(min_value=0, max_value=2**40),
        ),
        max_size=30,
    ),
    garbage=st.text(max_size=80),
)
def test_bpf_jit_sum_total(tmp_path_factory, rows, garbage):
    """vmallocinfo summing: totals exactly the bpf_jit rows, survives
    arbitrary garbage lines."""
    from gpud_amd.components.host.memory import read_bpf_jit_buffer_bytes

    p = tmp_path_factory.mktemp("vmi") / "vmallocinfo"
    lines = []
    expect = 0
    for tag, size in rows:
        if tag.startswith("bpf_jit"):
            expect += size
        lines.append(f"0xdead-0xbeef {size} {tag} pages=1")
    lines.append(garbage.replace("\n", " "))
    p.write_text("\n".join(lines) + "\n")
    assert read_bpf_jit_buffer_bytes(str(p)) == expect


@settings(max_examples=120, deadline=None)
@given(text=st.text(max_size=2000))
def test_lspci_acs_parser_total(text):
    from gpud_amd.components.host.pci import parse_acs_bridges

    out = parse_acs_bridges(text)
    assert isinstance(out, list)
