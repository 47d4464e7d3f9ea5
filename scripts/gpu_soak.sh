#!/usr/bin/env bash
# Soak: daemon on live HW at a 1 s poll interval for ~90 s with periodic
# fault injection through the API; verifies stability + flat RSS.
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo
timeout 380 python - > gpurun_out/soak.log 2>&1 <<'PYEOF'
import json, os, signal, socket, subprocess, sys, time
import psutil

s = socket.socket(); s.bind(("127.0.0.1", 0)); port = s.getsockname()[1]; s.close()
env = {**os.environ}
proc = subprocess.Popen(
    [sys.executable, "-m", "gpud_amd", "run", "--in-memory-db",
     "--address", f"127.0.0.1:{port}", "--log-level", "warning",
     "--poll-interval-seconds", "1"],
    start_new_session=True, env=env)
try:
    sys.path.insert(0, "/root/repo")
    from gpud_amd.client import Client
    c = Client(f"https://127.0.0.1:{port}")
    assert c.wait_healthz(60)
    p = psutil.Process(proc.pid)
    rss0 = p.memory_info().rss
    t_end = time.time() + 300
    i = 0
    diag_done = False
    import httpx as _hx
    long_client = _hx.Client(verify=False, timeout=240.0)
    while time.time() < t_end:
        # hammer the API + trigger fast checks; one full diag battery
        # (MFMA/GEMM/HBM/LDS, ~30-60 s synchronous) fires mid-soak
        states = c.get_health_states()
        c.trigger_check(component="accelerator-amd-temperature")
        c.trigger_check(component="accelerator-amd-ecc")
        if i % 10 == 0:
            try:
                c.inject_fault(ras_event_name="amdgpu_ring_timeout")
            except Exception:
                pass  # /dev/kmsg absent on some boxes -> expected 400
        if not diag_done and time.time() > t_end - 240:
            r = long_client.get(
                f"https://127.0.0.1:{port}/v1/components/trigger-check",
                params={"componentName": "accelerator-amd-diag-mfma"},
            )
            assert r.status_code == 200, r.text[:200]
            assert r.json()["states"][0]["health"] == "Healthy", r.text[:300]
            diag_done = True
        c.prometheus_metrics()
        i += 1
        time.sleep(1.0)
    assert diag_done
    rss1 = p.memory_info().rss
    cpu = p.cpu_percent(interval=2.0)
    states = c.get_health_states()
    summary = {
        "iterations": i,
        "rss_start_mb": round(rss0/1e6, 1),
        "rss_end_mb": round(rss1/1e6, 1),
        "rss_growth_mb": round((rss1-rss0)/1e6, 1),
        "daemon_cpu_percent_final": cpu,
        "components": len(states),
        "daemon_alive": proc.poll() is None,
    }
    print(json.dumps(summary))
    c.close()
finally:
    try: os.killpg(proc.pid, signal.SIGTERM)
    except ProcessLookupError: pass
    try: proc.wait(timeout=10)
    except subprocess.TimeoutExpired:
        os.killpg(proc.pid, signal.SIGKILL)
PYEOF
tail -3 gpurun_out/soak.log
