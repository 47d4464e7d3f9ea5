"""Link-state history store: drop/flap detection with a fake clock,
restart persistence, tombstone, retention (reference:
components/accelerator/nvidia/infiniband/store — store.go:57-319,
scan_drops.go, scan_flaps.go)."""

import pytest

from gpud_amd.pkg.link_store import (
    EVENT_DROP,
    EVENT_FLAP,
    LinkStore,
    STATE_ACTIVE,
    STATE_DOWN,
)
from gpud_amd.pkg.sqlite_util import open_memory_pair


class Clock:
    def __init__(self, t=1_700_000_000.0):
        self.t = t

    def __call__(self):
        return self.t

    def advance(self, s):
        self.t += s


@pytest.fixture()
def store_and_clock(tmp_path):
    clock = Clock()
    db_rw, db_ro = open_memory_pair()
    st = LinkStore(
        db_rw, db_ro, table_prefix="t", get_time_now=clock,
        min_insert_interval=15.0,
    )
    yield st, clock
    db_rw.close()


def _sweep(st, clock, state, advance=60.0):
    st.insert([{"device": "gpu0", "port": 0, "state": state}])
    clock.advance(advance)


def test_insert_rate_limited(store_and_clock):
    st, clock = store_and_clock
    assert st.insert([{"device": "d", "port": 1, "state": STATE_ACTIVE}])
    clock.advance(5)  # < 15s min interval
    assert not st.insert([{"device": "d", "port": 1, "state": STATE_ACTIVE}])
    clock.advance(15)
    assert st.insert([{"device": "d", "port": 1, "state": STATE_ACTIVE}])


def test_drop_detection_requires_persistent_down(store_and_clock):
    st, clock = store_and_clock
    # down for 2 minutes then recovered: below the 4-minute drop threshold
    _sweep(st, clock, STATE_DOWN)
    _sweep(st, clock, STATE_DOWN)
    _sweep(st, clock, STATE_ACTIVE)
    assert st.scan() == []
    # down for 5 consecutive minutes: drop
    for _ in range(6):
        _sweep(st, clock, STATE_DOWN)
    events = st.scan()
    assert any(e.event_type == EVENT_DROP for e in events)
    ev = st.evaluate()
    assert len(ev["drops"]) == 1
    assert "gpu0 port 0 down since" in ev["drops"][0].reason


def test_drop_sticky_window_then_autoclear(store_and_clock):
    st, clock = store_and_clock
    for _ in range(6):
        _sweep(st, clock, STATE_DOWN)
    _sweep(st, clock, STATE_ACTIVE)
    # recovered, but within the 10-minute sticky window: still surfaced
    assert len(st.evaluate(drop_sticky_window=600.0)["drops"]) == 1
    # stably recovered past the window: auto-cleared
    clock.advance(700)
    st.insert([{"device": "gpu0", "port": 0, "state": STATE_ACTIVE}])
    assert st.evaluate(drop_sticky_window=600.0)["drops"] == []


def test_flap_detection_needs_threshold_reverts(store_and_clock):
    st, clock = store_and_clock
    # two down->active reverts (each down span 60s >= 25s interval): below
    # the 3-revert threshold
    for _ in range(2):
        _sweep(st, clock, STATE_DOWN)
        _sweep(st, clock, STATE_DOWN)
        _sweep(st, clock, STATE_ACTIVE)
    assert all(e.event_type != EVENT_FLAP for e in st.scan())
    # a third revert crosses it
    _sweep(st, clock, STATE_DOWN)
    _sweep(st, clock, STATE_DOWN)
    _sweep(st, clock, STATE_ACTIVE)
    flaps = [e for e in st.scan() if e.event_type == EVENT_FLAP]
    assert len(flaps) == 3
    # sticky by default (flap window 0 = until set-healthy)
    clock.advance(24 * 3600 / 24)
    assert len(st.evaluate()["flaps"]) == 1  # one representative per port


def test_flap_auto_clear_window(store_and_clock):
    st, clock = store_and_clock
    for _ in range(3):
        _sweep(st, clock, STATE_DOWN)
        _sweep(st, clock, STATE_DOWN)
        _sweep(st, clock, STATE_ACTIVE)
    assert st.evaluate(flap_auto_clear_window=600.0)["flaps"]
    clock.advance(700)
    assert st.evaluate(flap_auto_clear_window=600.0)["flaps"] == []


def test_tombstone_clears_and_survives_restart(tmp_path):
    from gpud_amd.pkg.sqlite_util import open_ro, open_rw

    path = str(tmp_path / "state.db")
    clock = Clock()
    db_rw = open_rw(path)
    db_ro = open_ro(path)
    st = LinkStore(db_rw, db_ro, table_prefix="t", get_time_now=clock,
                   min_insert_interval=0.0)
    for _ in range(6):
        st.insert([{"device": "g", "port": 2, "state": STATE_DOWN}])
        clock.advance(60)
    assert st.evaluate()["drops"]
    # operator set-healthy -> tombstone; findings cleared
    st.set_tombstone()
    assert st.evaluate()["drops"] == []
    db_rw.close()
    db_ro.close()

    # daemon restart: NEW store over the same file — tombstone persists,
    # history persists
    db_rw = open_rw(path)
    db_ro = open_ro(path)
    st2 = LinkStore(db_rw, db_ro, table_prefix="t", get_time_now=clock,
                    min_insert_interval=0.0)
    assert st2.get_tombstone() > 0
    assert st2.evaluate()["drops"] == []
    # but NEW drops after the tombstone are found again
    clock.advance(60)
    for _ in range(6):
        st2.insert([{"device": "g", "port": 2, "state": STATE_DOWN}])
        clock.advance(60)
    assert st2.evaluate()["drops"]
    db_rw.close()
    db_ro.close()


def test_purge_retention(store_and_clock):
    st, clock = store_and_clock
    _sweep(st, clock, STATE_ACTIVE)
    clock.advance(st.retention_seconds + 100)
    _sweep(st, clock, STATE_ACTIVE)
    removed = st.purge()
    assert removed == 1


def test_xgmi_expected_links_by_product():
    from gpud_amd.components.accelerator.xgmi import (
        expected_links_for_product,
    )

    assert expected_links_for_product("AMD Instinct MI355X", 8) == 7
    assert expected_links_for_product("AMD Instinct MI355X", 4) == 3
    assert expected_links_for_product("AMD Instinct MI355X", 1) == 0
    assert expected_links_for_product("SomeOther GPU", 8) == 0


def test_xgmi_component_uses_store(monkeypatch, tmp_path):
    """Flapping xGMI links surface through the SQLite history and clear via
    set-healthy (tombstone)."""
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    monkeypatch.setenv("GPUD_AMDSMI_MOCK_GPUS", "2")
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config

    core = build_core(
        Config(data_dir=str(tmp_path)), in_memory_db=True,
        kmsg_writable=False, record_reboot=False,
    )
    try:
        comp = core.registry.get("accelerator-amd-xgmi")
        assert comp is not None and comp.link_store is not None
        clock = Clock()
        comp.link_store.now = clock
        comp.link_store.min_insert_interval = 0.0
        # drive 3 down->up flap cycles through the injected snapshot seam
        down = {"u1": {"xgmi_link_status": {"states": [0, 1]}},
                "u2": {"xgmi_link_status": {"states": [1, 1]}}}
        up = {"u1": {"xgmi_link_status": {"states": [1, 1]}},
              "u2": {"xgmi_link_status": {"states": [1, 1]}}}
        for _ in range(3):
            comp.get_snapshots = lambda: down
            comp.check()
            clock.advance(60)
            comp.check()
            clock.advance(60)
            comp.get_snapshots = lambda: up
            cr = comp.check()
            clock.advance(60)
        assert cr.health in ("Degraded", "Unhealthy")
        assert "flap" in cr.reason.lower() or "drop" in cr.reason.lower()
        # set-healthy tombstones the history
        comp.set_healthy()
        cr = comp.check()
        assert cr.health == "Healthy", cr.reason
    finally:
        core.close()


def test_infiniband_component_uses_store(monkeypatch, tmp_path):
    """IB port drops surface through the SQLite history store and persist
    until recovery + sticky window (the reference's ibports store)."""
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config

    core = build_core(
        Config(data_dir=str(tmp_path)), in_memory_db=True,
        kmsg_writable=False, record_reboot=False,
    )
    try:
        from gpud_amd.components.host.infiniband import InfinibandComponent

        comp = InfinibandComponent(core.gpud_instance)
        assert comp.link_store is not None
        clock = Clock()
        comp.link_store.now = clock
        comp.link_store.min_insert_interval = 0.0

        def ports(active):
            return [{
                "device": "mlx5_0", "port": "1", "active": active,
                "rate_gbps": 400.0,
                "counters": {"link_downed": 0 if active else 1,
                             "link_error_recovery": 0, "symbol_error": 0,
                             "port_rcv_errors": 0},
            }]

        # port down for 5+ minutes -> drop detected via the store
        comp.get_ports = lambda: ports(False)
        for _ in range(6):
            cr = comp.check()
            clock.advance(60)
        assert cr.health == "Unhealthy"
        # port recovers: still surfaced within the 10-min sticky window
        comp.get_ports = lambda: ports(True)
        cr = comp.check()
        assert cr.health in ("Unhealthy", "Degraded"), cr.reason
        assert "drop" in cr.reason.lower() or "flap" in cr.reason.lower()
        # set-healthy tombstones the finding
        comp.set_healthy()
        cr = comp.check()
        assert cr.health == "Healthy", cr.reason
    finally:
        core.close()
