"""Registry-wide component invariants (reference pattern: every component
satisfies the same interface contract — components/types.go Component).

Runs every registered component once under the mock backend and asserts
the contracts the server/session layers rely on: serializable health
states, queryable events, reference-style metric labeling, tag/name
hygiene. A new component that violates any of these breaks here rather
than in production."""

import datetime

import pytest

from gpud_amd.apiv1.types import HealthStateType, utcnow


@pytest.fixture(scope="module")
def core():
    import os

    os.environ["GPUD_AMDSMI_MOCK"] = "1"
    os.environ["GPUD_AMDSMI_MOCK_GPUS"] = "2"
    from gpud_amd.bootstrap import build_core

    c = build_core(in_memory_db=True, kmsg_writable=False, record_reboot=False)
    yield c
    c.close()
    os.environ.pop("GPUD_AMDSMI_MOCK", None)
    os.environ.pop("GPUD_AMDSMI_MOCK_GPUS", None)


def _non_diag(core):
    return [c for c in core.registry.all_components() if "diag" not in c.name]


def test_every_component_checks_healthy_under_mock(core):
    for comp in _non_diag(core):
        cr = comp.trigger_check()
        assert cr.health in (
            HealthStateType.HEALTHY,
            HealthStateType.DEGRADED,
        ), f"{comp.name}: {cr.health} — {cr.reason} / {cr.error}"


def test_every_component_serializes_states(core):
    for comp in core.registry.all_components():
        states = comp.last_health_states()
        assert states, comp.name
        for st in states:
            d = st.to_dict()
            assert d.get("component_name") or d.get("name") or d.get("component"), (
                comp.name,
                d,
            )
            assert d["health"] in ("Healthy", "Degraded", "Unhealthy", "Initializing")


def test_every_component_events_queryable(core):
    since = utcnow() - datetime.timedelta(hours=1)
    for comp in core.registry.all_components():
        evs = comp.events(since)
        assert evs is None or isinstance(evs, list), comp.name


def test_every_component_tags_include_name(core):
    names = set()
    for comp in core.registry.all_components():
        assert comp.name not in names, f"duplicate name {comp.name}"
        names.add(comp.name)
        assert comp.name in comp.tags(), comp.name
        assert comp.name == comp.name.lower(), comp.name


def test_all_gpud_metrics_carry_component_label(core):
    """The /v1/metrics scraper keeps only samples with the gpud_component
    label (reference: pkg/metrics/types.go:9) — every gauge a component
    registers must carry it or it silently vanishes from the API."""
    for comp in _non_diag(core):
        comp.trigger_check()
    for family in core.metrics_registry.collect():
        if family.name in ("process", "python_gc"):  # client defaults absent
            continue
        for sample in family.samples:
            if sample.name.startswith(("accelerator_amd_", "gpud_")) or any(
                sample.name.startswith(p)
                for p in ("cpu_", "memory_", "disk_", "os_", "network_")
            ):
                assert "gpud_component" in sample.labels, (
                    family.name,
                    sample.name,
                    sample.labels,
                )


def test_component_check_is_reentrant(core):
    """trigger_check twice in a row must not corrupt cached state."""
    for comp in _non_diag(core):
        a = comp.trigger_check()
        b = comp.trigger_check()
        assert b.health in ("Healthy", "Degraded", "Unhealthy"), comp.name
        assert type(a) is type(b)
