"""Core framework tests: api types, component model, registry, stores.

Mirrors the reference test strategy: injected function fields,
in-memory sqlite, no GPU required.
"""

import datetime
import json
import time

import pytest

from gpud_amd.apiv1.types import (
    Event,
    EventType,
    HealthState,
    HealthStateType,
    Metric,
    SuggestedActions,
    parse_rfc3339,
    rfc3339,
    utcnow,
)
from gpud_amd.components.base import (
    CheckResult,
    GPUdInstance,
    Registry,
    TickerComponent,
)
from gpud_amd.pkg import metadata
from gpud_amd.pkg.eventstore import Store, default_table_name
from gpud_amd.pkg.sqlite_util import compact, open_memory_pair, read_db_size


# ---------------------------------------------------------------------------
# api/v1 types
# ---------------------------------------------------------------------------

def test_health_state_roundtrip():
    hs = HealthState(
        component="cpu",
        name="cpu",
        health=HealthStateType.UNHEALTHY,
        reason="too hot",
        error="boom",
        suggested_actions=SuggestedActions(
            description="reboot it",
            repair_actions=["REBOOT_SYSTEM"],
        ),
        extra_info={"k": "v"},
    )
    d = hs.to_dict()
    assert d["component"] == "cpu"
    assert d["health"] == "Unhealthy"
    assert d["suggested_actions"]["repair_actions"] == ["REBOOT_SYSTEM"]
    # wire format keys match reference api/v1/types.go json tags
    assert "suggested_actions" in d and "extra_info" in d
    back = HealthState.from_dict(json.loads(json.dumps(d)))
    assert back.component == "cpu"
    assert back.suggested_actions.repair_actions == ["REBOOT_SYSTEM"]


def test_event_type_from_string():
    assert EventType.from_string("Info") == "Info"
    assert EventType.from_string("bogus") == "Unknown"


def test_rfc3339_roundtrip():
    t = datetime.datetime(2026, 1, 2, 3, 4, 5, tzinfo=datetime.timezone.utc)
    assert rfc3339(t) == "2026-01-02T03:04:05Z"
    assert parse_rfc3339("2026-01-02T03:04:05Z") == t


def test_health_state_omits_empty():
    d = HealthState(component="x", reason="").to_dict()
    assert "reason" not in d
    assert "error" not in d
    assert "suggested_actions" not in d


# ---------------------------------------------------------------------------
# component model
# ---------------------------------------------------------------------------

class FakeComponent(TickerComponent):
    poll_interval = 0.05

    def __init__(self, fail=False):
        super().__init__()
        self.fail = fail
        self.checks = 0

    @property
    def name(self):
        return "fake"

    def check(self):
        self.checks += 1
        if self.fail:
            raise RuntimeError("injected failure")
        return CheckResult(self.name, reason="ok")


def test_ticker_component_caches_result():
    c = FakeComponent()
    states = c.last_health_states()
    assert states[0].health == HealthStateType.INITIALIZING
    c.trigger_check()
    states = c.last_health_states()
    assert states[0].health == HealthStateType.HEALTHY
    assert states[0].reason == "ok"
    c.close()


def test_ticker_component_survives_exception():
    c = FakeComponent(fail=True)
    cr = c.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert "injected failure" in cr.error
    c.close()


def test_ticker_component_background_loop():
    c = FakeComponent()
    durations = []
    c.set_duration_observer(lambda name, d: durations.append((name, d)))
    c.start()
    deadline = time.time() + 2.0
    while c.checks < 2 and time.time() < deadline:
        time.sleep(0.01)
    c.close()
    assert c.checks >= 2
    assert durations and durations[0][0] == "fake"


def test_registry_register_get_deregister():
    reg = Registry(GPUdInstance())
    c = reg.register(lambda inst: FakeComponent())
    assert reg.get("fake") is c
    assert [x.name for x in reg.all_components()] == ["fake"]
    with pytest.raises(ValueError):
        reg.register(lambda inst: FakeComponent())
    # FakeComponent is not deregisterable
    assert reg.deregister("fake") is None
    assert reg.get("fake") is c


# ---------------------------------------------------------------------------
# sqlite + eventstore + metadata
# ---------------------------------------------------------------------------

def test_event_table_name_matches_reference_format():
    # reference: pkg/eventstore/database.go:136-143
    assert (
        default_table_name("accelerator-amd-error-ras")
        == "components_accelerator_amd_error_ras_events_v0_5_0"
    )
    assert default_table_name("os") == "components_os_events_v0_5_0"


def test_eventstore_insert_get_latest_purge(mem_db):
    rw, ro = mem_db
    store = Store(rw, ro, retention=datetime.timedelta(days=3))
    bucket = store.bucket("os", disable_purge=True)
    now = utcnow()
    old = now - datetime.timedelta(days=10)
    bucket.insert(Event(time=old, name="reboot", type="Warning", message="old"))
    bucket.insert(Event(time=now, name="reboot", type="Warning", message="new"))
    evs = bucket.get(now - datetime.timedelta(days=30))
    assert len(evs) == 2
    assert evs[0].message == "new"  # DESC order
    assert bucket.latest().message == "new"
    found = bucket.find(Event(time=now, name="reboot", type="Warning"))
    assert found is not None
    n = bucket.purge(int((now - datetime.timedelta(days=3)).timestamp()))
    assert n == 1
    assert len(bucket.get(now - datetime.timedelta(days=30))) == 1
    store.close()


def test_metadata_kv(mem_db):
    rw, ro = mem_db
    metadata.create_table(rw)
    metadata.set_value(rw, metadata.KEY_MACHINE_ID, "m-123")
    assert metadata.get_value(ro, metadata.KEY_MACHINE_ID) == "m-123"
    metadata.set_value(rw, metadata.KEY_MACHINE_ID, "m-456")
    assert metadata.get_value(ro, metadata.KEY_MACHINE_ID) == "m-456"
    assert metadata.all_values(ro) == {"machine_id": "m-456"}
    metadata.delete_value(rw, metadata.KEY_MACHINE_ID)
    assert metadata.get_value(ro, metadata.KEY_MACHINE_ID) == ""


def test_db_size_and_compact(mem_db):
    rw, _ = mem_db
    rw.execute("CREATE TABLE t (x INTEGER)")
    rw.execute("INSERT INTO t VALUES (1)")
    assert read_db_size(rw) > 0
    compact(rw)


# ---------------------------------------------------------------------------
# metrics pipeline
# ---------------------------------------------------------------------------

def test_metrics_pipeline_scrape_store_read(mem_db):
    from prometheus_client import Gauge

    from gpud_amd.pkg.metrics import (
        LABEL_COMPONENT,
        MetricsStore,
        Scraper,
        Syncer,
        create_registry,
    )

    rw, ro = mem_db
    reg = create_registry()
    g = Gauge(
        "accelerator_amd_temperature_current_celsius",
        "temp",
        [LABEL_COMPONENT, "uuid"],
        registry=reg,
    )
    g.labels(**{LABEL_COMPONENT: "accelerator-amd-temperature", "uuid": "gpu0"}).set(55)
    # a metric WITHOUT the gpud_component label must be filtered out
    Gauge("random_thirdparty_metric", "x", registry=reg).set(1)

    scraper = Scraper(reg)
    scraped = scraper.scrape()
    assert len(scraped) == 1
    assert scraped[0].component == "accelerator-amd-temperature"
    assert scraped[0].labels == {"uuid": "gpu0"}

    store = MetricsStore(rw, ro)
    syncer = Syncer(scraper, store, sync_interval_seconds=3600)
    n = syncer.sync_once()
    assert n == 1
    by_comp = store.read()
    assert "accelerator-amd-temperature" in by_comp
    m = by_comp["accelerator-amd-temperature"][0]
    assert m.value == 55
    assert m.labels == {"uuid": "gpu0"}

    # purge removes old rows
    removed = store.purge(datetime.datetime.now(datetime.timezone.utc) + datetime.timedelta(days=1))
    assert removed == 1


def test_recorder_self_telemetry(mem_db):
    from gpud_amd.pkg.metrics import Recorder, Scraper, create_registry

    rw, _ = mem_db
    reg = create_registry()
    rec = Recorder(reg, db_rw=rw, interval_seconds=3600)
    rec.record_once()
    rec.observe_check_duration("cpu", 0.01)
    scraped = Scraper(reg).scrape()
    names = {m.name for m in scraped}
    assert "gpud_state_db_size_bytes" in names
    assert "gpud_component_check_duration_seconds_count" in names
    comp_labels = {m.component for m in scraped}
    assert "gpud" in comp_labels and "cpu" in comp_labels


def test_eventstore_find_by_name_since_bounds(mem_db):
    """Name filter + since bound are both honored, and the returned order
    is newest-first (the API contract /v1/events relies on)."""
    import datetime

    from gpud_amd.apiv1.types import Event, utcnow
    from gpud_amd.pkg.eventstore import Store

    rw, ro = mem_db
    store = Store(rw, ro)
    b = store.bucket("bounds", disable_purge=True)
    now = utcnow()
    for mins, name in [(50, "a"), (40, "b"), (30, "a"), (20, "b"), (10, "a")]:
        b.insert(
            Event(
                time=now - datetime.timedelta(minutes=mins),
                name=name,
                type="Info",
                message=f"{name}@{mins}",
            )
        )
    got = b.find_by_name_since(name="a", since=now - datetime.timedelta(minutes=35))
    assert [e.message for e in got] == ["a@10", "a@30"]
    got = b.find_by_name_since(name="b", since=now - datetime.timedelta(minutes=999))
    assert [e.message for e in got] == ["b@20", "b@40"]
    assert b.find_by_name_since(name="zzz", since=now - datetime.timedelta(days=1)) == []
    store.close()


def test_metrics_store_component_and_since_filters(mem_db):
    import datetime

    from gpud_amd.pkg.metrics.scraper import ScrapedMetric
    from gpud_amd.pkg.metrics.store import MetricsStore

    rw, ro = mem_db
    store = MetricsStore(rw, ro)
    now = datetime.datetime.now(datetime.timezone.utc)
    old = now - datetime.timedelta(hours=2)

    def m(comp, name, value, ts):
        return ScrapedMetric(
            unix_ms=int(ts.timestamp() * 1000),
            component=comp,
            name=name,
            labels={},
            value=value,
        )

    store.record([m("cpu", "cpu_usage", 10.0, old),
                  m("cpu", "cpu_usage", 20.0, now),
                  m("memory", "memory_used", 5.0, now)])
    everything = store.read()
    assert set(everything) == {"cpu", "memory"}
    only_cpu = store.read(components=["cpu"])
    assert set(only_cpu) == {"cpu"} and len(only_cpu["cpu"]) == 2
    recent = store.read(since=now - datetime.timedelta(minutes=30))
    assert len(recent["cpu"]) == 1 and recent["cpu"][0].value == 20.0
    both = store.read(since=now - datetime.timedelta(minutes=30),
                      components=["memory"])
    assert set(both) == {"memory"}


def test_eventstore_drops_legacy_v0_4_0_table(mem_db):
    """A reference-era v0_4_0 table in a carried-over gpud.state is
    dropped at bucket open (reference: eventstore/database.go:97-103)."""
    from gpud_amd.pkg.eventstore import Store, default_table_name

    rw, ro = mem_db
    legacy = default_table_name("error-ras", "v0_4_0")
    rw.executescript(f"CREATE TABLE {legacy} (timestamp INTEGER);")
    store = Store(rw, ro)
    store.bucket("error-ras", disable_purge=True)
    rows = ro.query(
        "SELECT name FROM sqlite_master WHERE type='table' AND name=?",
        (legacy,),
    )
    assert rows == []
    store.close()
