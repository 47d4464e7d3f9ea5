#!/usr/bin/env bash
# Round-2 extended soak on the shipped state: daemon at a 1 s poll
# interval for 30 min with API hammering, periodic fault injection, a
# full diag battery (incl. the v7P GEMM) every ~8 min, an RSS series,
# and cores enabled (the round-1 heap corruption is root-caused and
# mutex-guarded; this run proves the final build stays flat).
set -x
cd /root/repo
mkdir -p gpurun_out
ulimit -c unlimited
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo
timeout 2000 python - > gpurun_out/soak_r2.log 2>&1 <<'PYEOF'
import json, os, signal, socket, subprocess, sys, time
import psutil

s = socket.socket(); s.bind(("127.0.0.1", 0)); port = s.getsockname()[1]; s.close()
proc = subprocess.Popen(
    [sys.executable, "-m", "gpud_amd", "run", "--in-memory-db",
     "--address", f"127.0.0.1:{port}", "--log-level", "warning",
     "--poll-interval-seconds", "1"],
    start_new_session=True, env={**os.environ})
try:
    sys.path.insert(0, "/root/repo")
    from gpud_amd.client import Client
    c = Client(f"https://127.0.0.1:{port}")
    assert c.wait_healthz(60)
    p = psutil.Process(proc.pid)
    rss0 = p.memory_info().rss
    rss_series = []
    DURATION = 1800
    t0 = time.time()
    t_end = t0 + DURATION
    i = 0
    diag_runs = 0
    next_diag = t0 + 120
    import httpx as _hx
    long_client = _hx.Client(verify=False, timeout=300.0)
    while time.time() < t_end:
        c.get_health_states()
        c.trigger_check(component="accelerator-amd-temperature")
        c.trigger_check(component="accelerator-amd-ecc")
        c.trigger_check(component="accelerator-amd-xgmi")
        if i % 10 == 0:
            try:
                c.inject_fault(ras_event_name="amdgpu_ring_timeout")
            except Exception:
                pass  # /dev/kmsg write-limited on some boxes
        if i % 60 == 0:
            rss_series.append(
                (round(time.time() - t0), round(p.memory_info().rss / 1e6, 1)))
        if time.time() >= next_diag and time.time() < t_end - 300:
            r = long_client.get(
                f"https://127.0.0.1:{port}/v1/components/trigger-check",
                params={"componentName": "accelerator-amd-diag-mfma"},
            )
            assert r.status_code == 200, r.text[:200]
            st = r.json()["states"][0]
            assert st["health"] == "Healthy", r.text[:300]
            gemm = (st.get("extra_info") or {}).get("gpu0.gemm_bf16_tflops")
            print(json.dumps({"diag_at_s": round(time.time() - t0),
                              "gemm_bf16_tflops": gemm}), flush=True)
            diag_runs += 1
            next_diag = time.time() + 480
        c.prometheus_metrics()
        i += 1
        time.sleep(1.0)
    rss1 = p.memory_info().rss
    summary = {
        "duration_s": DURATION,
        "iterations": i,
        "diag_batteries": diag_runs,
        "rss_start_mb": round(rss0 / 1e6, 1),
        "rss_end_mb": round(rss1 / 1e6, 1),
        "rss_growth_mb": round((rss1 - rss0) / 1e6, 1),
        "rss_series": rss_series,
        "daemon_cpu_percent_final": p.cpu_percent(interval=2.0),
        "components": len(c.get_health_states()),
        "daemon_alive": proc.poll() is None,
        "cores_in_cwd": [f for f in os.listdir(".") if f.startswith("core")],
    }
    assert diag_runs >= 2, diag_runs
    assert proc.poll() is None
    print(json.dumps(summary))
    c.close()
finally:
    try: os.killpg(proc.pid, signal.SIGTERM)
    except ProcessLookupError: pass
    try: proc.wait(timeout=10)
    except subprocess.TimeoutExpired:
        os.killpg(proc.pid, signal.SIGKILL)
PYEOF
echo "soak rc=$?" >> gpurun_out/soak_r2.log
tail -6 gpurun_out/soak_r2.log
