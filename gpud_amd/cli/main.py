"""gpud CLI (reference: cmd/gpud/command/command.go:60-936 subcommands:
up/down/run/scan/status/compact/inject-fault/set-healthy/machine-info/
metadata/list-plugins/custom-plugins/run-plugin-group/login/logout)."""

from __future__ import annotations

import json
import os
import signal
import sys
import time
import typer

from .. import __version__
from ..pkg.config import Config, DEFAULT_DATA_DIR
from ..pkg.log import setup as log_setup

app = typer.Typer(name="gpud-amd", help="MI355X-native GPU health daemon")


def _load_config(
    data_dir: str,
    address: str,
    expected_gpu_count: int = 0,
    plugin_specs_file: str = "",
    endpoint: str = "",
) -> Config:
    cfg = Config(data_dir=data_dir)
    if address:
        cfg.address = address
    cfg.expected_gpu_count = expected_gpu_count
    cfg.plugin_specs_file = plugin_specs_file
    cfg.endpoint = endpoint
    return cfg


@app.command()
def version():
    """Print the gpud-amd version."""
    typer.echo(f"gpud-amd {__version__}")


@app.command()
def run(
    data_dir: str = typer.Option(DEFAULT_DATA_DIR, help="state directory"),
    address: str = typer.Option("localhost:15132", help="listen address"),
    expected_gpu_count: int = typer.Option(0),
    expected_xgmi_link_count: int = typer.Option(
        0, help="xGMI links per GPU that must be UP (7 on an 8-OAM node)"
    ),
    expected_ib_ports: int = typer.Option(0),
    expected_ib_rate_gbps: float = typer.Option(0.0),
    expected_compute_partition: str = typer.Option(
        "", help="alert if compute partition mode differs (e.g. SPX)"
    ),
    expected_memory_partition: str = typer.Option(
        "", help="alert if memory partition mode differs (e.g. NPS1)"
    ),
    events_retention_days: float = typer.Option(
        3.0, help="days to retain component events (reference: "
        "events-retention-period)"
    ),
    compact_period_hours: float = typer.Option(24.0),
    components: str = typer.Option(
        "", help="comma-separated allow-list of components to enable"
    ),
    disabled_components: str = typer.Option(
        "", help="comma-separated components to disable"
    ),
    kernel_modules_to_check: str = typer.Option(
        "", help="comma-separated kernel modules that must be loaded"
    ),
    temperature_margin_celsius: float = typer.Option(
        10.0, help="degraded when a temp is within this margin of its limit "
        "(reference: threshold-celsius-slowdown-margin)"
    ),
    ras_reboot_threshold: int = typer.Option(
        2, help="reboots tolerated before RAS errors escalate to hardware "
        "inspection (reference: xid-reboot-threshold)"
    ),
    session_protocol: str = typer.Option(
        "auto", help="control-plane session protocol: v1 (HTTP dual-stream), "
        "v2 (gRPC, reference protobuf wire format), or auto (v2 with "
        "fallback to v1 when the manager reports it unsupported)"
    ),
    ras_event_thresholds: str = typer.Option(
        "", help='per-event escalation overrides as JSON, e.g. '
        '\'{"amdgpu_ring_timeout": 1}\' (reference: xid-thresholds)'
    ),
    poll_interval_seconds: float = typer.Option(60.0),
    plugin_specs_file: str = typer.Option(""),
    endpoint: str = typer.Option("", help="control-plane endpoint"),
    token: str = typer.Option("", help="control-plane token"),
    control_plane_ca_file: str = typer.Option(
        "", help="CA bundle to verify the control-plane endpoint against "
        "(default: system CAs)"
    ),
    control_plane_insecure_tls: bool = typer.Option(
        False, help="DANGEROUS: skip TLS verification towards the control "
        "plane (lab use only; the session carries the auth token and "
        "accepts bootstrap/update requests)"
    ),
    in_memory_db: bool = typer.Option(False, help="ephemeral state (testing)"),
    kmsg_path: str = typer.Option(
        "/dev/kmsg", help="kmsg source/sink; a regular file switches the "
        "fault-injection loop to a file seam (rate-limited-kmsg hosts)"
    ),
    no_tls: bool = typer.Option(False, help="serve plain HTTP (testing)"),
    log_level: str = typer.Option("info"),
    # SMI-level failure injection for exercising the daemon on healthy
    # hardware (reference: gpud run --gpu-uuids-with-* flags,
    # cmd/gpud/run/command.go:272-335); comma-separated GPU uuids
    gpu_uuids_with_gpu_lost: str = typer.Option("", help="inject: GPU lost"),
    gpu_uuids_with_requires_reset: str = typer.Option(""),
    gpu_uuids_with_bad_page_pending: str = typer.Option(
        "", help="inject: bad pages pending (row-remapping-pending analog)"
    ),
    gpu_uuids_with_bad_page_threshold: str = typer.Option(
        "", help="inject: bad-page count at threshold (row-remapping-failed analog)"
    ),
    gpu_uuids_with_throttle: str = typer.Option(
        "", help="inject: power throttling (hw-slowdown analog)"
    ),
    gpu_uuids_with_thermal_throttle: str = typer.Option(
        "", help="inject: thermal throttling (hw-slowdown-thermal analog)"
    ),
    gpu_uuids_with_xgmi_unhealthy: str = typer.Option(
        "", help="inject: xGMI link down (fabric-state analog)"
    ),
    gpu_uuids_with_ecc_uncorrectable: str = typer.Option(
        "", help="inject: uncorrectable ECC"
    ),
):
    """Run the daemon (reference: cmd/gpud/run/command.go:42)."""
    log_setup(level=log_level)
    from ..bootstrap import build_core
    from ..pkg import custom_plugins
    from ..pkg.fault_injector import SMIFailureInjector
    from ..server import Server

    def _uuid_set(v: str):
        return {u.strip() for u in v.split(",") if u.strip()}

    injector = SMIFailureInjector(
        gpu_lost_uuids=_uuid_set(gpu_uuids_with_gpu_lost),
        requires_reset_uuids=_uuid_set(gpu_uuids_with_requires_reset),
        bad_page_pending_uuids=_uuid_set(gpu_uuids_with_bad_page_pending),
        bad_page_threshold_uuids=_uuid_set(gpu_uuids_with_bad_page_threshold),
        throttle_uuids=_uuid_set(gpu_uuids_with_throttle),
        thermal_throttle_uuids=_uuid_set(gpu_uuids_with_thermal_throttle),
        xgmi_unhealthy_uuids=_uuid_set(gpu_uuids_with_xgmi_unhealthy),
        ecc_uncorrectable_uuids=_uuid_set(gpu_uuids_with_ecc_uncorrectable),
    )

    cfg = _load_config(
        data_dir, address, expected_gpu_count, plugin_specs_file, endpoint
    )
    cfg.token = token
    cfg.control_plane_ca_file = control_plane_ca_file
    cfg.control_plane_insecure_tls = control_plane_insecure_tls
    cfg.expected_xgmi_link_count = expected_xgmi_link_count
    cfg.expected_compute_partition = expected_compute_partition
    cfg.expected_memory_partition = expected_memory_partition
    cfg.events_retention_days = events_retention_days
    cfg.compact_period_hours = compact_period_hours
    if components:
        cfg.enabled_components = [c.strip() for c in components.split(",") if c.strip()]
    if disabled_components:
        cfg.disabled_components = [
            c.strip() for c in disabled_components.split(",") if c.strip()
        ]
    if kernel_modules_to_check:
        cfg.kernel_modules_to_check = [
            m.strip() for m in kernel_modules_to_check.split(",") if m.strip()
        ]
    cfg.temperature_margin_threshold_c = temperature_margin_celsius
    cfg.ras_reboot_threshold = ras_reboot_threshold
    if ras_event_thresholds:
        try:
            cfg.ras_event_thresholds = {
                str(k): int(v)
                for k, v in json.loads(ras_event_thresholds).items()
            }
        except (ValueError, AttributeError) as e:
            typer.echo(f"invalid --ras-event-thresholds: {e}", err=True)
            raise typer.Exit(code=1)
    cfg.expected_ib_ports = expected_ib_ports
    cfg.expected_ib_rate_gbps = expected_ib_rate_gbps
    cfg.poll_interval_seconds = poll_interval_seconds
    cfg.kmsg_path = kmsg_path
    if not in_memory_db:
        os.makedirs(cfg.data_dir, exist_ok=True)

    specs = []
    if plugin_specs_file and os.path.exists(plugin_specs_file):
        specs = custom_plugins.load_specs(plugin_specs_file)
        err = custom_plugins.run_init_plugins(specs)
        if err:
            typer.echo(f"init plugin failure: {err}", err=True)
            raise typer.Exit(code=1)

    core = build_core(
        cfg, in_memory_db=in_memory_db, smi_failure_injector=injector
    )

    # token FIFO (reference: pkg/server/server.go:640-710 — a named pipe at
    # <dataDir>/gpud.fifo other local processes write a fresh CP token into)
    if not in_memory_db:
        try:
            if not os.path.exists(cfg.fifo_path):
                os.mkfifo(cfg.fifo_path, 0o600)
            # owner-only regardless of umask or a pre-existing pipe: any
            # wider mode lets an unprivileged local user inject a
            # control-plane token
            os.chmod(cfg.fifo_path, 0o600)

            def _fifo_watch():
                from ..pkg import metadata as _md

                while True:
                    try:
                        with open(cfg.fifo_path) as f:  # blocks for a writer
                            new_token = f.read().strip()
                        if new_token:
                            _md.set_value(core.db_rw, _md.KEY_TOKEN, new_token)
                    except OSError:
                        return

            import threading as _th

            _th.Thread(target=_fifo_watch, daemon=True).start()
        except OSError:
            pass
    # register plugin components
    for spec in specs:
        if spec.plugin_type == custom_plugins.PLUGIN_TYPE_INIT:
            continue
        for comp in custom_plugins.make_components(spec):
            try:
                core.registry.register_component(comp)
            except ValueError:
                pass
    core.start_components()
    core.start_background()

    host, _, port = cfg.address.rpartition(":")
    server = Server(
        core,
        host=host or "127.0.0.1",
        port=int(port),
        tls=not no_tls,
        plugin_specs=specs,
    )
    server.start()
    typer.echo(f"gpud-amd serving on {server.base_url}")

    session = None
    if endpoint:
        # optional login on boot (reference: cmd/gpud/run/command.go:99-127)
        if token:
            from ..pkg.login import do_login

            err = do_login(cfg, token=token, endpoint=endpoint)
            if err:
                typer.echo(f"login failed (continuing): {err}", err=True)
        from ..pkg import metadata as _md
        from ..session import Session

        machine_id = ""
        try:
            machine_id = _md.get_value(core.db_ro, _md.KEY_MACHINE_ID)
        except Exception:
            pass
        if session_protocol in ("v2", "auto"):
            # gRPC bidi stream in the reference's protobuf wire format
            # (reference: --session-protocol, cmd/gpud/run/command.go:156)
            from ..session.v2 import V2Session

            dispatcher = Session(
                core,
                endpoint=endpoint,
                token=token,
                machine_id=machine_id,
                open_reader=lambda: iter(()),
                send_response=lambda f: None,
            )
            grpc_endpoint = (
                endpoint.replace("https://", "").replace("http://", "")
            )
            fallback = None
            if session_protocol == "auto":
                # reference ProtocolAuto: permanent fallback to the legacy
                # session on UNIMPLEMENTED (session_keepalive.go:15)
                def fallback(
                    _core=core, _ep=endpoint, _tok=token, _mid=machine_id
                ):
                    legacy = Session(
                        _core, endpoint=_ep, token=_tok, machine_id=_mid
                    )
                    legacy.start()

            v2_creds = None
            if cfg.control_plane_ca_file and not cfg.control_plane_insecure_tls:
                import grpc as _grpc

                with open(cfg.control_plane_ca_file, "rb") as _caf:
                    v2_creds = _grpc.ssl_channel_credentials(
                        root_certificates=_caf.read()
                    )
            session = V2Session(
                dispatcher,
                endpoint=grpc_endpoint,
                machine_id=machine_id,
                token=token,
                on_unsupported=fallback,
                credentials=v2_creds,
                insecure=cfg.control_plane_insecure_tls,
            )
        else:
            session = Session(
                core, endpoint=endpoint, token=token, machine_id=machine_id
            )
        session.start()

    # package manager reconcile loops (reference: cmd/gpud/run:425-431)
    from ..pkg.gpud_manager import PackageController

    pkg_controller = PackageController(cfg)
    core.pkg_controller = pkg_controller  # /admin/packages reads live status
    pkg_controller.start()

    stop = {"flag": False}

    # version-file-triggered update+exit (reference: server.go:815
    # updateFromVersionFile — systemd Restart=always picks up the new build)
    def _version_watch():
        from ..pkg.update import check_version_file

        while not stop["flag"]:
            time.sleep(60.0)
            try:
                pending = check_version_file(cfg)
            except Exception:
                continue
            if pending:
                typer.echo(
                    f"target version {pending} requested; exiting for restart",
                    err=True,
                )
                stop["flag"] = True
                return

    import threading as _threading

    _threading.Thread(target=_version_watch, daemon=True).start()

    # amdsmi-appears-later restart (reference: pkg/nvidia/nvml/
    # instance.go:121-147 refreshNVMLAndExit — a daemon booted before the
    # GPU driver exits cleanly once the library loads, so systemd's
    # Restart=always brings it back with the accelerator components live)
    if core.smi_instance is not None and not core.smi_instance.exists:
        def _smi_watch():
            from .. import smi as _smi

            while not stop["flag"]:
                time.sleep(60.0)
                try:
                    probe = _smi.new()
                except Exception:
                    continue
                if probe.exists:
                    probe.shutdown()
                    typer.echo(
                        "amdsmi became available; exiting for restart", err=True
                    )
                    stop["flag"] = True
                    return

        _threading.Thread(target=_smi_watch, daemon=True).start()

    def _sig(_s, _f):
        stop["flag"] = True

    signal.signal(signal.SIGTERM, _sig)
    signal.signal(signal.SIGINT, _sig)
    try:
        while not stop["flag"]:
            time.sleep(0.5)
    finally:
        pkg_controller.stop()
        if session is not None:
            session.stop()
        server.stop()
        core.close()


@app.command()
@app.command("check", hidden=True)  # reference alias: scan == check == s
@app.command("s", hidden=True)
def scan(
    expected_gpu_count: int = typer.Option(0),
    mock: bool = typer.Option(False, help="use the mock SMI backend"),
    output: str = typer.Option("table", help="table | json"),
):
    """One-shot health scan, no daemon/DB (reference: pkg/scan/scan.go:33)."""
    log_setup(level="warning")
    if mock:
        os.environ["GPUD_AMDSMI_MOCK"] = "1"
    from ..apiv1.types import HealthStateType
    from ..bootstrap import build_core

    cfg = Config()
    cfg.expected_gpu_count = expected_gpu_count
    core = build_core(
        cfg, in_memory_db=True, kmsg_writable=False, record_reboot=False
    )
    try:
        rows = []
        worst = HealthStateType.HEALTHY
        for comp in core.registry.all_components():
            if not comp.is_supported():
                rows.append((comp.name, "-", "not supported on this host", None))
                continue
            if getattr(comp, "run_mode", "") == "manual":
                rows.append((comp.name, "-", "manual run mode (diag)", None))
                continue
            cr = comp.trigger_check()
            h = cr.health_state_type()
            rows.append((comp.name, h, cr.summary()[:90], cr))
            if h == HealthStateType.UNHEALTHY:
                worst = h
            elif h == HealthStateType.DEGRADED and worst == HealthStateType.HEALTHY:
                worst = h
        if output == "json":
            out = {
                "overall": worst,
                "components": [
                    {
                        "component": name,
                        "health": health,
                        "reason": reason,
                        **(
                            {"states": [st.to_dict() for st in cr.health_states()]}
                            if cr is not None
                            else {}
                        ),
                    }
                    for name, health, reason, cr in rows
                ],
            }
            typer.echo(json.dumps(out, indent=1))
        else:
            width = max(len(r[0]) for r in rows)
            for name, health, reason, _cr in rows:
                mark = {"Healthy": "✔", "Degraded": "~", "Unhealthy": "✘"}.get(health, " ")
                typer.echo(f"{mark} {name:<{width}}  {health:<10} {reason}")
            typer.echo(f"\noverall: {worst}")
        raise typer.Exit(code=0 if worst != HealthStateType.UNHEALTHY else 1)
    finally:
        core.close()


@app.command()
def diagnose(
    output: str = typer.Option("", help="write the JSON report here"),
    rocprof: bool = typer.Option(
        False, help="also capture a rocprofv3 kernel trace of the MFMA stress"
    ),
    quick: bool = typer.Option(False, help="shorter kernel runs"),
):
    """Run the active CDNA4 diagnostics (MFMA/HBM/LDS/fabric) and report
    per-GPU numbers — the DCGM-diag analog."""
    log_setup(level="warning")
    from ..bootstrap import build_core

    core = build_core(
        Config(), in_memory_db=True, kmsg_writable=False, record_reboot=False
    )
    try:
        report = {"diagnostics": {}}
        for name in (
            "accelerator-amd-diag-mfma",
            "accelerator-amd-diag-bandwidth",
            "accelerator-amd-diag-fabric",
        ):
            comp = core.registry.get(name)
            if comp is None:
                continue
            if quick and hasattr(comp, "iters"):
                comp.iters = 512
            if quick and hasattr(comp, "gemm_iters"):
                comp.gemm_iters = 1
            cr = comp.trigger_check()
            report["diagnostics"][name] = {
                "health": cr.health,
                "reason": cr.reason,
                "error": cr.error,
                "measurements": cr.extra_info or {},
            }
            mark = {"Healthy": "✔", "Unhealthy": "✘"}.get(cr.health, "~")
            typer.echo(f"{mark} {name}: {cr.health} — {cr.reason[:100]}")
        if rocprof:
            import shutil as _sh
            import subprocess as _sp
            import tempfile as _tf

            if _sh.which("rocprofv3"):
                d = _tf.mkdtemp(prefix="gpud-rocprof-")
                res = _sp.run(
                    [
                        "rocprofv3", "--kernel-trace", "--stats", "-d", d,
                        "--", sys.executable, "-c",
                        "from gpud_amd.diag import _diag; _diag.set_device(0);"
                        "print(_diag.mfma_stress_bf16(iters=512, workgroups=1024))",
                    ],
                    capture_output=True, text=True, timeout=300,
                    cwd=os.path.dirname(os.path.dirname(os.path.dirname(
                        os.path.abspath(__file__)))),
                )
                report["rocprof"] = {
                    "output_dir": d,
                    "rc": res.returncode,
                    "stdout_tail": res.stdout[-1500:],
                }
                typer.echo(f"rocprof trace written under {d}")
            else:
                report["rocprof"] = {"error": "rocprofv3 not on PATH"}
        overall = all(
            v["health"] == "Healthy" for v in report["diagnostics"].values()
        )
        report["overall"] = "Healthy" if overall else "Unhealthy"
        if output:
            with open(output, "w") as f:
                json.dump(report, f, indent=1)
            typer.echo(f"report written to {output}")
        raise typer.Exit(code=0 if overall else 1)
    finally:
        core.close()


@app.command()
def status(
    server_url: str = typer.Option("https://localhost:15132"),
    watch: float = typer.Option(0.0, help="re-poll every N seconds"),
    data_dir: str = typer.Option(
        DEFAULT_DATA_DIR, help="state dir (for the session-login history line)"
    ),
):
    """Query a running daemon's health states (reference: gpud status)."""
    from ..client import Client

    # control-plane session history (reference: pkg/session/states)
    try:
        from ..pkg import session_states as _ss
        from ..pkg.sqlite_util import open_ro as _open_ro

        conn = _open_ro(Config(data_dir=data_dir).state_path)
        last = _ss.read_last(conn)
        conn.close()
        if last is not None:
            mark = "ok" if last.success else "FAILED"
            typer.echo(
                f"control-plane session: last login {mark} ({last.message})"
            )
    except Exception:  # noqa: BLE001 — no state file yet is fine
        pass

    c = Client(server_url)
    if not c.wait_healthz(timeout=5):
        typer.echo("daemon not reachable", err=True)
        raise typer.Exit(code=1)
    try:
        while True:
            states = c.get_health_states()
            for comp, sts in sorted(states.items()):
                for s in sts:
                    typer.echo(f"{comp:45s} {s.health:<12} {s.reason[:80]}")
            if watch <= 0:
                break
            typer.echo("-" * 80)
            time.sleep(watch)
    except KeyboardInterrupt:
        pass
    finally:
        c.close()


@app.command()
def compact(
    data_dir: str = typer.Option(DEFAULT_DATA_DIR),
):
    """Offline VACUUM of the state DB (reference: gpud compact)."""
    from ..pkg.sqlite_util import compact as do_compact, open_rw, read_db_size

    cfg = Config(data_dir=data_dir)
    conn = open_rw(cfg.state_path)
    before = read_db_size(conn)
    do_compact(conn)
    after = read_db_size(conn)
    conn.close()
    typer.echo(f"compacted {cfg.state_path}: {before} -> {after} bytes")


@app.command("inject-fault")
def inject_fault(
    server_url: str = typer.Option("https://localhost:15132"),
    ras_event: str = typer.Option("", help="catalog event name to inject"),
    kernel_message: str = typer.Option("", help="raw kernel message"),
):
    """Inject a synthetic fault into the running daemon."""
    from ..client import Client

    c = Client(server_url)
    out = c.inject_fault(ras_event_name=ras_event, kernel_message=kernel_message)
    typer.echo(json.dumps(out))
    c.close()


@app.command("set-healthy")
def set_healthy(
    server_url: str = typer.Option("https://localhost:15132"),
    components: str = typer.Option("", help="comma-separated component names"),
):
    from ..client import Client

    c = Client(server_url)
    out = c.set_healthy(components.split(",") if components else None)
    typer.echo(json.dumps(out))
    c.close()


@app.command()
def bundle(
    output: str = typer.Option("", help="output tar.gz path"),
    mock: bool = typer.Option(False, help="use the mock SMI backend"),
):
    """Collect a support bundle (the nvidia-bug-report.sh analog):
    amdsmi state, kernel messages, component health, host context."""
    log_setup(level="warning")
    if mock:
        os.environ["GPUD_AMDSMI_MOCK"] = "1"
    import time as _time

    from ..bootstrap import build_core
    from ..pkg.bundle import collect_bundle

    out = output or f"gpud-bundle-{_time.strftime('%Y%m%d-%H%M%S')}.tar.gz"
    core = build_core(
        Config(), in_memory_db=True, kmsg_writable=False, record_reboot=False
    )
    try:
        for comp in core.registry.all_components():
            try:
                comp.trigger_check()
            except Exception:  # noqa: BLE001 — bundle what we can
                pass
        path = collect_bundle(out, core=core)
        typer.echo(f"bundle written: {path} ({os.path.getsize(path)} bytes)")
    finally:
        core.close()


@app.command("machine-info")
def machine_info(mock: bool = typer.Option(False)):
    if mock:
        os.environ["GPUD_AMDSMI_MOCK"] = "1"
    from .. import smi
    from ..pkg.machine_info import get_machine_info

    inst = smi.new()
    typer.echo(json.dumps(get_machine_info(inst).to_dict(), indent=2))
    inst.shutdown()


@app.command()
def metadata(data_dir: str = typer.Option(DEFAULT_DATA_DIR)):
    """Inspect the metadata table (reference: gpud metadata)."""
    from ..pkg import metadata as md
    from ..pkg.sqlite_util import open_ro

    cfg = Config(data_dir=data_dir)
    conn = open_ro(cfg.state_path)
    for k, v in md.all_values(conn).items():
        shown = v if k not in (md.KEY_TOKEN, md.KEY_MACHINE_PROOF) else "***"
        typer.echo(f"{k}\t{shown}")
    conn.close()


@app.command("list-plugins")
def list_plugins(specs_file: str = typer.Argument(...)):
    from ..pkg import custom_plugins

    for spec in custom_plugins.load_specs(specs_file):
        typer.echo(
            f"{spec.plugin_name}\t{spec.plugin_type}\t{spec.run_mode}\t"
            f"every {spec.interval_seconds:g}s"
        )


@app.command("custom-plugins")
@app.command("plugins", hidden=True)  # reference aliases: cs/plugin/plugins
def custom_plugins_cmd(
    specs_file: str = typer.Argument(...),
    run: bool = typer.Option(False, "--run", "-r", help="run the plugins"),
    fail_fast: bool = typer.Option(
        True, "--fail-fast/--no-fail-fast", "-f", help="exit on first unhealthy"
    ),
):
    """Check (and optionally run) a custom-plugin specs file
    (reference: gpud custom-plugins — cmd/gpud/command/command.go:805)."""
    from ..pkg import custom_plugins

    try:
        specs = custom_plugins.load_specs(specs_file)
    except Exception as e:  # noqa: BLE001 — surface any parse error as exit 1
        typer.echo(f"invalid plugin specs: {e}", err=True)
        raise typer.Exit(code=1)
    typer.echo(f"valid plugin specs: {len(specs)} plugin(s)")
    if not run:
        for spec in specs:
            typer.echo(f"  {spec.plugin_name} ({spec.plugin_type}, {spec.run_mode})")
        raise typer.Exit(code=0)
    failed = 0
    for spec in specs:
        for comp in custom_plugins.make_components(spec):
            cr = comp.trigger_check()
            typer.echo(f"{comp.name}: {cr.health} ({cr.reason})")
            if cr.health != "Healthy":
                failed += 1
                if fail_fast:
                    raise typer.Exit(code=1)
    raise typer.Exit(code=1 if failed else 0)


@app.command("run-plugin-group")
def run_plugin_group(
    specs_file: str = typer.Argument(...),
    tag: str = typer.Option("", help="only plugins with this tag"),
):
    """Run custom plugins once, locally (reference: gpud run-plugin-group)."""
    from ..pkg import custom_plugins

    failed = 0
    for spec in custom_plugins.load_specs(specs_file):
        if tag and tag not in spec.tags:
            continue
        for comp in custom_plugins.make_components(spec):
            cr = comp.trigger_check()
            typer.echo(f"{comp.name}: {cr.health} ({cr.reason})")
            if cr.health != "Healthy":
                failed += 1
    raise typer.Exit(code=1 if failed else 0)


@app.command()
def notify(
    event: str = typer.Argument(..., help="startup | shutdown"),
    data_dir: str = typer.Option(DEFAULT_DATA_DIR),
):
    """Notify the control plane of daemon startup/shutdown
    (reference: gpud notify)."""
    import httpx

    from ..pkg import metadata as md
    from ..pkg.sqlite_util import open_ro

    cfg = Config(data_dir=data_dir)
    try:
        conn = open_ro(cfg.state_path)
        endpoint = md.get_value(conn, md.KEY_ENDPOINT)
        machine_id = md.get_value(conn, md.KEY_MACHINE_ID)
        token = md.get_value(conn, md.KEY_TOKEN)
        conn.close()
    except Exception as e:
        typer.echo(f"cannot read metadata: {e}", err=True)
        raise typer.Exit(code=1)
    if not endpoint:
        typer.echo("no control-plane endpoint configured; nothing to notify")
        raise typer.Exit(code=0)
    try:
        r = httpx.post(
            endpoint.rstrip("/") + "/api/v1/notify",
            json={"machineID": machine_id, "event": event},
            headers={"token": token},
            timeout=10,
            verify=cfg.control_plane_verify(),
        )
        typer.echo(f"notify {event}: HTTP {r.status_code}")
    except httpx.HTTPError as e:
        typer.echo(f"notify failed: {e}", err=True)
        raise typer.Exit(code=1)


@app.command("update-check")
def update_check(data_dir: str = typer.Option(DEFAULT_DATA_DIR)):
    """Check whether the version file requests a new version
    (reference: gpud update check — cmd/gpud/command/command.go:453)."""
    from ..pkg.update import check_version_file

    cfg = Config(data_dir=data_dir)
    pending = check_version_file(cfg)
    if pending:
        typer.echo(f"update available: {pending}")
        raise typer.Exit(code=0)
    typer.echo("up to date")


@app.command()
def update(
    version: str = typer.Argument(...),
    data_dir: str = typer.Option(DEFAULT_DATA_DIR),
    base_url: str = typer.Option("https://pkg.gpud.dev/packages"),
):
    """Self-update to a version (reference: gpud update)."""
    from ..pkg.update import update_to_version

    cfg = Config(data_dir=data_dir)
    err = update_to_version(cfg, version, base_url=base_url)
    if err:
        typer.echo(f"update failed: {err}", err=True)
        raise typer.Exit(code=1)
    typer.echo(f"updated to {version}")


@app.command()
def login(
    token: str = typer.Argument(...),
    endpoint: str = typer.Option("https://api.gpud.ai"),
    data_dir: str = typer.Option(DEFAULT_DATA_DIR),
    node_group: str = typer.Option(""),
):
    """Log in to the control plane and persist credentials."""
    from ..pkg.login import do_login

    cfg = Config(data_dir=data_dir)
    os.makedirs(cfg.data_dir, exist_ok=True)
    err = do_login(cfg, token=token, endpoint=endpoint, node_group=node_group)
    if err:
        typer.echo(f"login failed: {err}", err=True)
        raise typer.Exit(code=1)
    typer.echo("login ok")


@app.command()
def logout(data_dir: str = typer.Option(DEFAULT_DATA_DIR)):
    """Clear control-plane credentials (reference: gpud logout)."""
    from ..pkg import metadata as md
    from ..pkg.sqlite_util import open_rw

    cfg = Config(data_dir=data_dir)
    conn = open_rw(cfg.state_path)
    md.create_table(conn)
    md.delete_value(conn, md.KEY_TOKEN)
    md.delete_value(conn, md.KEY_MACHINE_PROOF)
    conn.close()
    typer.echo("logged out")


release_app = typer.Typer(help="release signing (reference: pkg/release/distsign)")
app.add_typer(release_app, name="release")


@release_app.command("gen-key")
def release_gen_key(out_prefix: str = typer.Argument(...)):
    """Generate an ed25519 keypair: <prefix>.key (seed) + <prefix>.pub."""
    from ..pkg import distsign

    seed, pub = distsign.generate_keypair()
    with open(out_prefix + ".key", "wb") as f:
        f.write(seed)
    os.chmod(out_prefix + ".key", 0o600)
    with open(out_prefix + ".pub", "wb") as f:
        f.write(pub)
    typer.echo(f"wrote {out_prefix}.key and {out_prefix}.pub")


@release_app.command("sign")
def release_sign(
    artifact: str = typer.Argument(...),
    key: str = typer.Option(..., help="path to the .key seed file"),
):
    """Sign an artifact; writes <artifact>.sig."""
    from ..pkg import distsign

    with open(artifact, "rb") as f:
        data = f.read()
    with open(key, "rb") as f:
        seed = f.read()
    sig = distsign.sign(data, seed)
    with open(artifact + ".sig", "wb") as f:
        f.write(sig)
    typer.echo(f"wrote {artifact}.sig")


@release_app.command("verify")
def release_verify(
    artifact: str = typer.Argument(...),
    pub: str = typer.Option(..., help="path to the .pub key file"),
    sig: str = typer.Option("", help="signature file (default <artifact>.sig)"),
):
    from ..pkg import distsign

    with open(artifact, "rb") as f:
        data = f.read()
    with open(sig or artifact + ".sig", "rb") as f:
        signature = f.read()
    with open(pub, "rb") as f:
        public = f.read()
    if distsign.verify(data, signature, public):
        typer.echo("signature OK")
    else:
        typer.echo("signature INVALID", err=True)
        raise typer.Exit(code=1)


@app.command()
def up(
    data_dir: str = typer.Option(DEFAULT_DATA_DIR),
    token: str = typer.Option(""),
    endpoint: str = typer.Option(""),
):
    """Install + start the systemd service (reference: cmd/gpud/up)."""
    from ..pkg.systemd_util import install_and_start

    err = install_and_start(data_dir=data_dir, token=token, endpoint=endpoint)
    if err:
        typer.echo(err, err=True)
        raise typer.Exit(code=1)
    typer.echo("gpud-amd systemd service installed and started")


@app.command()
def down():
    """Stop + disable the systemd service (reference: cmd/gpud down)."""
    from ..pkg.systemd_util import stop_and_disable

    err = stop_and_disable()
    if err:
        typer.echo(err, err=True)
        raise typer.Exit(code=1)
    typer.echo("gpud-amd systemd service stopped")


def main():
    app()


if __name__ == "__main__":
    main()
