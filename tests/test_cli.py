"""CLI tests via typer's CliRunner (reference: cmd/gpud command tests)."""

import json
import os

import pytest
from typer.testing import CliRunner

from gpud_amd.cli.main import app

runner = CliRunner()


@pytest.fixture(autouse=True)
def mock_smi(monkeypatch):
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    monkeypatch.setenv("GPUD_AMDSMI_MOCK_GPUS", "2")


def test_version():
    res = runner.invoke(app, ["version"])
    assert res.exit_code == 0
    assert "gpud-amd" in res.output


def test_scan_mock():
    res = runner.invoke(app, ["scan", "--mock"])
    assert res.exit_code == 0, res.output
    assert "accelerator-amd-temperature" in res.output
    assert "overall: Healthy" in res.output


def test_scan_mock_expected_count_mismatch():
    res = runner.invoke(app, ["scan", "--mock", "--expected-gpu-count", "8"])
    assert res.exit_code == 1
    assert "Unhealthy" in res.output


def test_machine_info_mock():
    res = runner.invoke(app, ["machine-info", "--mock"])
    assert res.exit_code == 0, res.output
    mi = json.loads(res.output)
    assert mi["gpuInfo"]["product"] == "AMD Instinct MI355X"


def test_compact_and_metadata(tmp_path):
    from gpud_amd.pkg import metadata
    from gpud_amd.pkg.config import Config
    from gpud_amd.pkg.sqlite_util import open_rw

    cfg = Config(data_dir=str(tmp_path))
    conn = open_rw(cfg.state_path)
    metadata.create_table(conn)
    metadata.set_value(conn, metadata.KEY_MACHINE_ID, "m-42")
    metadata.set_value(conn, metadata.KEY_TOKEN, "secret")
    conn.close()
    res = runner.invoke(app, ["compact", "--data-dir", str(tmp_path)])
    assert res.exit_code == 0 and "compacted" in res.output
    res = runner.invoke(app, ["metadata", "--data-dir", str(tmp_path)])
    assert res.exit_code == 0
    assert "m-42" in res.output
    assert "secret" not in res.output  # token redacted


def test_list_plugins_and_run_group(tmp_path):
    specs = tmp_path / "plugins.yaml"
    specs.write_text(
        """
- plugin_name: quick
  plugin_type: component
  tags: [grp]
  health_state_plugin:
    steps:
      - run_bash_script:
          script: echo fine
"""
    )
    res = runner.invoke(app, ["list-plugins", str(specs)])
    assert res.exit_code == 0 and "quick" in res.output
    res = runner.invoke(app, ["run-plugin-group", str(specs), "--tag", "grp"])
    assert res.exit_code == 0
    assert "custom-plugin-quick: Healthy" in res.output


def test_release_sign_verify(tmp_path):
    art = tmp_path / "artifact.bin"
    art.write_bytes(b"release payload")
    prefix = str(tmp_path / "rootkey")
    res = runner.invoke(app, ["release", "gen-key", prefix])
    assert res.exit_code == 0, res.output
    res = runner.invoke(app, ["release", "sign", str(art), "--key", prefix + ".key"])
    assert res.exit_code == 0, res.output
    res = runner.invoke(app, ["release", "verify", str(art), "--pub", prefix + ".pub"])
    assert res.exit_code == 0 and "OK" in res.output
    # tamper -> invalid
    art.write_bytes(b"tampered payload!")
    res = runner.invoke(app, ["release", "verify", str(art), "--pub", prefix + ".pub"])
    assert res.exit_code == 1


def test_logout_cli(tmp_path):
    from gpud_amd.pkg import metadata
    from gpud_amd.pkg.config import Config
    from gpud_amd.pkg.sqlite_util import open_ro, open_rw

    cfg = Config(data_dir=str(tmp_path))
    conn = open_rw(cfg.state_path)
    metadata.create_table(conn)
    metadata.set_value(conn, metadata.KEY_TOKEN, "tok")
    conn.close()
    res = runner.invoke(app, ["logout", "--data-dir", str(tmp_path)])
    assert res.exit_code == 0
    conn = open_ro(cfg.state_path)
    assert metadata.get_value(conn, metadata.KEY_TOKEN) == ""
    conn.close()


def test_scan_json_output():
    res = runner.invoke(app, ["scan", "--mock", "--output", "json"])
    assert res.exit_code == 0, res.output
    out = json.loads(res.output)
    assert out["overall"] == "Healthy"
    comps = {c["component"]: c for c in out["components"]}
    assert comps["accelerator-amd-temperature"]["health"] == "Healthy"
    assert "states" in comps["cpu"]


def test_example_plugins_load_and_run():
    """The shipped examples/plugins.yaml must stay valid and runnable."""
    import os

    path = os.path.join(
        os.path.dirname(__file__), "..", "examples", "plugins.yaml"
    )
    from gpud_amd.pkg.custom_plugins import load_specs, make_components, run_init_plugins

    specs = load_specs(path)
    assert [s.plugin_name for s in specs] == [
        "rocm-present",
        "sysfs-gpu-count",
        "mount-writable",
    ]
    assert run_init_plugins(specs) is None  # /opt/rocm exists here
    comps = [
        c
        for s in specs
        if s.plugin_type != "init"
        for c in make_components(s)
    ]
    names = [c.name for c in comps]
    assert "custom-plugin-mount-writable-tmp" in names
    # the tmp mount-writable probe should pass anywhere
    tmp_comp = next(c for c in comps if c.name.endswith("-tmp"))
    cr = tmp_comp.trigger_check()
    assert cr.health == "Healthy", cr.raw_output
    # the sysfs crosscheck parses its own JSON (health depends on whether
    # this host has AMD cards in sysfs)
    sysfs_comp = next(c for c in comps if c.name.endswith("gpu-count"))
    cr = sysfs_comp.trigger_check()
    assert "amd_cards" in (cr.extra_info or {}), cr.raw_output


def test_custom_plugins_validate_and_run(tmp_path):
    """Reference: gpud custom-plugins (validate; -r to run; fail-fast)."""
    from typer.testing import CliRunner

    from gpud_amd.cli.main import app

    spec = tmp_path / "plugins.yaml"
    spec.write_text(
        """
- plugin_name: ok-plugin
  plugin_type: component
  run_mode: auto
  health_state_plugin:
    steps:
      - name: run
        run_bash_script:
          content_type: plaintext
          script: "echo fine"
- plugin_name: bad-plugin
  plugin_type: component
  run_mode: auto
  health_state_plugin:
    steps:
      - name: run
        run_bash_script:
          content_type: plaintext
          script: "exit 3"
"""
    )
    runner = CliRunner()
    r = runner.invoke(app, ["custom-plugins", str(spec)])
    assert r.exit_code == 0
    assert "valid plugin specs: 2" in r.output
    # run with fail-fast (default): the bad plugin stops the run with code 1
    r = runner.invoke(app, ["custom-plugins", str(spec), "-r"])
    assert r.exit_code == 1
    assert "ok-plugin: Healthy" in r.output
    # invalid file ⇒ exit 1
    bad = tmp_path / "broken.yaml"
    bad.write_text("{not valid yaml: [")
    r = runner.invoke(app, ["custom-plugins", str(bad)])
    assert r.exit_code == 1


def test_update_check_pending(tmp_path):
    from typer.testing import CliRunner

    from gpud_amd.cli.main import app
    from gpud_amd.pkg.config import Config
    from gpud_amd.pkg.update import write_target_version

    cfg = Config(data_dir=str(tmp_path))
    runner = CliRunner()
    r = runner.invoke(app, ["update-check", "--data-dir", str(tmp_path)])
    assert r.exit_code == 0 and "up to date" in r.output
    write_target_version(cfg, "v99.0.0")
    r = runner.invoke(app, ["update-check", "--data-dir", str(tmp_path)])
    assert "update available: v99.0.0" in r.output


def test_scan_alias_check(monkeypatch):
    from typer.testing import CliRunner

    from gpud_amd.cli.main import app

    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    r = CliRunner().invoke(app, ["check", "--mock", "--output", "json"])
    assert r.exit_code == 0


def test_run_flags_map_to_config(monkeypatch):
    """The reference's key run flags have CLI analogs that reach Config
    (cmd/gpud/run/command.go flag surface)."""
    import subprocess
    import sys

    import gpud_amd.cli.main as cli_main

    captured = {}

    def fake_build_core(cfg, **kw):
        captured["cfg"] = cfg
        raise SystemExit(0)  # stop before the server boots

    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    monkeypatch.setattr("gpud_amd.bootstrap.build_core", fake_build_core)
    from typer.testing import CliRunner

    r = CliRunner().invoke(
        cli_main.app,
        [
            "run", "--in-memory-db",
            "--events-retention-days", "7",
            "--components", "cpu,memory",
            "--kernel-modules-to-check", "amdgpu,amdkfd",
            "--temperature-margin-celsius", "5",
            "--ras-reboot-threshold", "4",
            "--expected-compute-partition", "spx",
        ],
    )
    cfg = captured["cfg"]
    assert cfg.events_retention_days == 7
    assert cfg.enabled_components == ["cpu", "memory"]
    assert cfg.kernel_modules_to_check == ["amdgpu", "amdkfd"]
    assert cfg.temperature_margin_threshold_c == 5
    assert cfg.ras_reboot_threshold == 4
    assert cfg.expected_compute_partition == "spx"


def test_ras_reboot_threshold_reaches_component(monkeypatch, tmp_path):
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config

    cfg = Config(data_dir=str(tmp_path), ras_reboot_threshold=5)
    core = build_core(cfg, in_memory_db=True, kmsg_writable=False, record_reboot=False)
    try:
        comp = core.registry.get("accelerator-amd-error-ras")
        assert comp.reboot_threshold == 5
    finally:
        core.close()


def test_version_and_help_render():
    from typer.testing import CliRunner

    from gpud_amd import __version__
    from gpud_amd.cli.main import app

    r = CliRunner().invoke(app, ["version"])
    assert r.exit_code == 0 and __version__ in r.output
    r = CliRunner().invoke(app, ["--help"])
    assert r.exit_code == 0
    for cmd in ("run", "scan", "diagnose", "bundle", "custom-plugins",
                "inject-fault", "update-check", "up", "down", "release"):
        assert cmd in r.output, cmd
    r = CliRunner().invoke(app, ["run", "--help"], env={"COLUMNS": "250"})
    assert r.exit_code == 0
    flat = "".join(r.output.split())  # rich wraps long flags across lines
    for flag in ("--gpu-uuids-with-gpu-lost", "--session-protocol",
                 "--expected-compute-partition", "--ras-event-thresholds"):
        assert flag in flat, flag
