"""Host-component parser/logic tests with fixtures and injected getters."""

import pytest

from gpud_amd.apiv1.types import HealthStateType


@pytest.fixture()
def mock_core(monkeypatch, tmp_path):
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config

    core = build_core(
        Config(data_dir=str(tmp_path)),
        in_memory_db=True,
        kmsg_writable=False,
        record_reboot=False,
    )
    yield core
    core.close()


LSPCI_FIXTURE = """\
00:01.0 PCI bridge: Advanced Micro Devices, Inc. [AMD] Device 14a4
\tCapabilities: [2a0 v1] Access Control Services
\t\tACSCap: SrcValid+ TransBlk+ ReqRedir+ CmpltRedir+
\t\tACSCtl: SrcValid+ TransBlk- ReqRedir+ CmpltRedir+
00:03.0 PCI bridge: Advanced Micro Devices, Inc. [AMD] Device 14a4
\tCapabilities: [2a0 v1] Access Control Services
\t\tACSCap: SrcValid+ TransBlk+ ReqRedir+ CmpltRedir+
\t\tACSCtl: SrcValid- TransBlk- ReqRedir- CmpltRedir-
0a:00.0 Processing accelerators: Advanced Micro Devices [AMD] Device 75a0
\tCapabilities: [2a0 v1] Access Control Services
\t\tACSCtl: SrcValid+
"""


def test_pci_acs_parse():
    from gpud_amd.components.host.pci import parse_acs_bridges

    enabled = parse_acs_bridges(LSPCI_FIXTURE)
    # only the BRIDGE with SrcValid+ — never endpoints, never SrcValid-
    assert enabled == ["00:01.0"]


def test_pci_component_degraded_on_acs(mock_core):
    comp = mock_core.registry.get("pci")
    comp.get_virt_env = lambda: "none"  # ACS check is virt-gated
    comp.get_acs_bridges = lambda: ["00:01.0"]
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.DEGRADED
    comp.get_acs_bridges = lambda: []
    assert comp.trigger_check().health == HealthStateType.HEALTHY
    comp.get_acs_bridges = lambda: None
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY
    assert "skipped" in cr.reason


def test_fuse_connections_fixture(tmp_path, mock_core):
    from gpud_amd.components.host.fuse import read_connections

    root = tmp_path / "connections"
    for cid, waiting, maxbg in [("39", "2", "12"), ("40", "11", "12")]:
        d = root / cid
        d.mkdir(parents=True)
        (d / "waiting").write_text(waiting + "\n")
        (d / "max_background").write_text(maxbg + "\n")
    conns = read_connections(str(root))
    assert ("39", 2, 12) in conns and ("40", 11, 12) in conns
    comp = mock_core.registry.get("fuse")
    comp.get_connections = lambda: conns
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.DEGRADED  # 11/12 >= 90%
    assert "40" in cr.reason


def test_library_component_resolution(mock_core, tmp_path):
    comp = mock_core.registry.get("library")
    lib = tmp_path / "libfoo.so.1"
    lib.write_bytes(b"")
    comp.libraries = {"libfoo.so": [str(tmp_path)], "libmissing.so": []}
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert "libmissing.so" in cr.reason
    comp.libraries = {"libfoo.so": [str(tmp_path)]}
    assert comp.trigger_check().health == HealthStateType.HEALTHY


def test_network_latency_with_injected_probe(mock_core):
    comp = mock_core.registry.get("network-latency")
    comp.targets = [("edge-a", 443), ("edge-b", 443)]
    comp.probe = lambda h, p: 12.5 if h == "edge-a" else -1.0
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY
    assert cr.extra_info["edge-b:443"] == "unreachable"
    comp.probe = lambda h, p: -1.0
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY


def test_disk_usage_thresholds(mock_core, monkeypatch):
    import collections

    comp = mock_core.registry.get("disk")
    Usage = collections.namedtuple("Usage", "total used free percent")

    import gpud_amd.components.host.disk as diskmod

    monkeypatch.setattr(
        diskmod.psutil, "disk_usage", lambda mp: Usage(100, 99, 1, 99.0)
    )
    comp.mount_points = ["/"]
    comp.mount_targets = []
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    monkeypatch.setattr(
        diskmod.psutil, "disk_usage", lambda mp: Usage(100, 91, 9, 91.0)
    )
    assert comp.trigger_check().health == HealthStateType.DEGRADED
    monkeypatch.setattr(
        diskmod.psutil, "disk_usage", lambda mp: Usage(100, 10, 90, 10.0)
    )
    assert comp.trigger_check().health == HealthStateType.HEALTHY


def test_os_zombie_thresholds(mock_core):
    comp = mock_core.registry.get("os")
    comp.get_process_states = lambda: {"total": 5000, "zombies": 2500, "dstate": 0}
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    comp.get_process_states = lambda: {"total": 5000, "zombies": 1200, "dstate": 0}
    assert comp.trigger_check().health == HealthStateType.DEGRADED


def test_containerd_miss_threshold(mock_core):
    comp = mock_core.registry.get("containerd")
    comp.check_socket = lambda: False
    h1 = comp.trigger_check().health
    h2 = comp.trigger_check().health
    h3 = comp.trigger_check().health
    assert (h1, h2) == (HealthStateType.DEGRADED, HealthStateType.DEGRADED)
    assert h3 == HealthStateType.UNHEALTHY
    comp.check_socket = lambda: True
    comp.check_service = lambda: "active"
    assert comp.trigger_check().health == HealthStateType.HEALTHY


def test_systemd_util_paths(monkeypatch, tmp_path):
    import gpud_amd.pkg.systemd_util as su

    monkeypatch.setattr(su, "UNIT_PATH", str(tmp_path / "gpud-amd.service"))
    monkeypatch.setattr(su, "ENV_PATH", str(tmp_path / "gpud-amd.env"))
    monkeypatch.setattr(su.shutil, "which", lambda _: None)
    err = su.install_and_start(data_dir=str(tmp_path / "data"), endpoint="https://cp")
    assert err is not None and "systemctl" in err
    # the unit + env files were still written correctly
    unit = (tmp_path / "gpud-amd.service").read_text()
    assert "ExecStart=" in unit and "Restart=always" in unit
    assert "--endpoint https://cp" in (tmp_path / "gpud-amd.env").read_text()
    assert su.stop_and_disable() is not None
