"""gpud_amd flagship benchmark: the daemon poll cycle on real MI355X GPUs.

BASELINE.json metric: "daemon CPU% + p50 poll-cycle ms at 8 MI355X; diag
MFMA TFLOPS/GPU" — the reference (leptonai/gpud) publishes no numbers
(BASELINE.md), so this measures our daemon directly:

  * one "step" = one complete accelerator poll cycle for the GPUs this
    rank monitors: a native amdsmi telemetry sweep (temperature, power,
    clocks, activity, VRAM, ECC, throttle, xGMI, bad pages) followed by
    every accelerator component Check() evaluating health rules and
    updating Prometheus gauges — exactly what the daemon does per tick;
  * weak scaling: rank r monitors GPU r, so per-GPU work is fixed as N
    grows (the reference daemon monitors all GPUs from one process; flat
    scaling to 8 GPUs is the design goal — SURVEY.md §7);
  * headline value = p50 poll-cycle milliseconds (max over ranks);
  * extras: daemon CPU%% during the timed region and the per-GPU MFMA
    bf16 diag TFLOPS (measured after the timed region).

Launch: python bench.py --gpus N --steps K --warmup W
(N>1 via torch.distributed.run, one rank per GPU over RCCL).

--single-process measures the daemon's REAL deployment shape instead: ONE
process polls all N GPUs per cycle (the daemon monitors every GPU from one
process — SURVEY.md §7 "overhead flat to 8 GPUs"). With --mock it sets the
mock backend to N GPUs so the 8-GPU single-process cycle cost is measurable
on a CPU-only host; on a GPU box it monitors min(N, visible) live GPUs.
"""

import argparse
import json
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


class _RankFilteredSMI:
    """Expose only this rank's GPU through the Instance API so every
    component built on top of it (snapshots, process lists, link metrics)
    monitors exactly one GPU per rank — fixed per-GPU work as N grows."""

    def __init__(self, inst, index: int):
        self._inst = inst
        self.index = index
        uuids = inst.device_uuids()
        self.uuid = uuids[index % len(uuids)] if uuids else ""

    @property
    def failure_injector(self):
        return self._inst.failure_injector

    @failure_injector.setter
    def failure_injector(self, fi):
        self._inst.failure_injector = fi

    @property
    def exists(self):
        return self._inst.exists

    def init_error(self):
        return self._inst.init_error()

    def devices(self):
        devs = self._inst.devices()
        return {self.uuid: devs[self.uuid]} if self.uuid in devs else {}

    def device_uuids(self):
        return [self.uuid] if self.uuid else []

    def device_count(self):
        return 1 if self.uuid else 0

    @property
    def product_name(self):
        return self._inst.product_name

    def snapshot_all(self):
        from gpud_amd.smi import Instance

        devs = self._inst.devices()  # applies gpu-lost filtering
        if self.uuid not in devs:
            return {}
        snap = devs[self.uuid].snapshot()
        fi = self._inst.failure_injector
        if fi is not None:
            snap = Instance._apply_injection(self.uuid, snap, fi)
        return {self.uuid: snap}

    @property
    def driver_version(self):
        return self._inst.driver_version

    @property
    def rocm_version(self):
        return self._inst.rocm_version

    def shutdown(self):
        pass


class _SubsetSMI(_RankFilteredSMI):
    """Expose the FIRST n GPUs of the instance (single-process mode: one
    process monitors n GPUs per cycle, the daemon's real deployment shape)."""

    def __init__(self, inst, n: int):
        self._inst = inst
        uuids = inst.device_uuids()
        self._uuids = uuids[: max(1, n)]
        self.uuid = self._uuids[0] if self._uuids else ""

    def devices(self):
        devs = self._inst.devices()
        return {u: devs[u] for u in self._uuids if u in devs}

    def device_uuids(self):
        return list(self._uuids)

    def device_count(self):
        return len(self._uuids)

    def snapshot_all(self):
        from gpud_amd.smi import Instance

        devs = self._inst.devices()
        fi = self._inst.failure_injector
        out = {}
        for u in self._uuids:
            if u not in devs:
                continue
            snap = devs[u].snapshot()
            if fi is not None:
                snap = Instance._apply_injection(u, snap, fi)
            out[u] = snap
        return out


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=200)
    p.add_argument("--warmup", type=int, default=20)
    p.add_argument("--mock", action="store_true", help="CPU-only: mock SMI backend")
    p.add_argument(
        "--single-process",
        action="store_true",
        help="one process polls all --gpus GPUs per cycle (the daemon's real "
        "deployment shape) instead of one rank per GPU",
    )
    p.add_argument(
        "--fault-replay",
        action="store_true",
        help="inject SMI-level faults on a schedule during the timed region "
        "(BASELINE.json config 5) and report detection counts",
    )
    args = p.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if args.single_process:
        world = 1  # one process monitors all GPUs; n_gpus reported separately

    if args.mock:
        os.environ["GPUD_AMDSMI_MOCK"] = "1"
        if args.single_process:
            # mock exactly the requested node shape (e.g. 8 GPUs, 1 process)
            os.environ["GPUD_AMDSMI_MOCK_GPUS"] = str(max(1, args.gpus))

    if args.single_process and int(os.environ.get("WORLD_SIZE", "1")) > 1:
        print(
            json.dumps({"error": "--single-process is incompatible with torchrun"}),
            file=sys.stderr,
        )
        return 1

    import torch
    import torch.distributed as dist

    use_cuda = torch.cuda.is_available() and not args.mock
    if use_cuda:
        torch.cuda.set_device(local_rank % max(torch.cuda.device_count(), 1))

    distributed = world > 1
    if distributed:
        backend = "nccl" if use_cuda else "gloo"
        dist.init_process_group(backend=backend)

    from gpud_amd import smi as smi_pkg
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config
    import psutil

    cfg = Config()
    # filter to this rank's GPU BEFORE building the core so every component
    # is constructed against the single-GPU view
    inst = smi_pkg.new()
    if inst.exists and inst.device_count() > 0 and args.single_process:
        n_want = max(1, args.gpus)
        smi_for_core = _SubsetSMI(inst, n_want)
        n_gpus_seen = smi_for_core.device_count()
        data_source = "mock" if args.mock else "amdsmi"
    elif inst.exists and inst.device_count() > 0:
        smi_for_core = _RankFilteredSMI(inst, local_rank)
        n_gpus_seen = 1
        data_source = "mock" if args.mock else "amdsmi"
    else:
        if not args.mock:
            print(
                json.dumps(
                    {
                        "error": "no AMD GPU visible and --mock not set",
                        "init_error": inst.init_error(),
                    }
                ),
                file=sys.stderr,
            )
            return 1
        smi_for_core = inst
        n_gpus_seen = inst.device_count()
        data_source = "mock"
    core = build_core(
        cfg,
        in_memory_db=True,
        smi_instance=smi_for_core,
        kmsg_writable=False,
        record_reboot=False,
    )

    accel_components = [
        c
        for c in core.registry.all_components()
        if c.name.startswith("accelerator-amd-")
        and "diag" not in c.name
    ]

    # fault-replay: flip a rotating SMI-level failure on/off every 10 cycles
    # and count how many cycles observe a non-Healthy accelerator state —
    # the injected-fault load the north star's baseline asks for.
    fi = core.smi_failure_injector
    my_uuids = core.gpud_instance.smi.device_uuids()
    fault_kinds = ("ecc", "throttle", "xgmi", "bad_pages")
    replay = {"cycle": 0, "active": "", "detected": 0, "injected_cycles": 0}

    def _apply_fault(kind: str, on: bool) -> None:
        if not my_uuids:
            return
        u = my_uuids[0]
        target = {
            "ecc": fi.ecc_uncorrectable_uuids,
            "throttle": fi.throttle_uuids,
            "xgmi": fi.xgmi_unhealthy_uuids,
            "bad_pages": fi.bad_page_pending_uuids,
        }[kind]
        (target.add if on else target.discard)(u)

    def one_cycle() -> None:
        if args.fault_replay:
            c_i = replay["cycle"]
            if c_i % 10 == 0:
                if replay["active"]:
                    _apply_fault(replay["active"], False)
                replay["active"] = (
                    fault_kinds[(c_i // 10) % len(fault_kinds)]
                    if (c_i // 10) % 2 == 0
                    else ""
                )
                if replay["active"]:
                    _apply_fault(replay["active"], True)
            replay["cycle"] = c_i + 1
        core.shared_snapshots.refresh()
        unhealthy_seen = False
        for c in accel_components:
            cr = c.trigger_check()
            if cr.health != "Healthy":
                unhealthy_seen = True
        if args.fault_replay and replay["active"]:
            replay["injected_cycles"] += 1
            if unhealthy_seen:
                replay["detected"] += 1

    def barrier_sync() -> None:
        if distributed:
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    proc = psutil.Process()

    # warmup
    for _ in range(args.warmup):
        one_cycle()

    barrier_sync()
    cpu0 = proc.cpu_times()
    wall0 = time.monotonic()
    cycle_ms = []
    for _ in range(args.steps):
        t0 = time.perf_counter()
        one_cycle()
        cycle_ms.append((time.perf_counter() - t0) * 1000.0)
    barrier_sync()
    wall1 = time.monotonic()
    cpu1 = proc.cpu_times()

    wall = wall1 - wall0
    cpu_used = (cpu1.user - cpu0.user) + (cpu1.system - cpu0.system)
    cpu_pct = 100.0 * cpu_used / wall if wall > 0 else 0.0
    p50 = statistics.median(cycle_ms)
    mean_ms = sum(cycle_ms) / len(cycle_ms)
    p99 = sorted(cycle_ms)[max(0, int(len(cycle_ms) * 0.99) - 1)]

    # diag MFMA TFLOPS (after the timed region; real GPU only)
    mfma_tflops = None
    if use_cuda:
        try:
            from gpud_amd.diag import _diag

            _diag.set_device(local_rank % max(torch.cuda.device_count(), 1))
            res = _diag.mfma_stress_bf16(iters=1024, workgroups=1024)
            if res["verified"]:
                mfma_tflops = round(float(res["tflops"]), 1)
        except Exception:
            mfma_tflops = None

    # MAX over ranks for the time metrics (the whole-job aggregate)
    if distributed:
        t = torch.tensor([p50, mean_ms, cpu_pct, p99], dtype=torch.float64)
        if use_cuda:
            t = t.cuda()
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        p50, mean_ms, cpu_pct_max, p99 = [float(x) for x in t.tolist()]
    else:
        cpu_pct_max = cpu_pct

    if rank == 0:
        out = {
            "metric": "poll_cycle_p50_ms",
            "value": round(p50, 4),
            "unit": "ms",
            "n_gpus": n_gpus_seen if args.single_process else world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(mean_ms, 4),
            "higher_is_better": False,
            "scaling": "weak",
            "vs_baseline": None,
            # telemetry poll metric: no compute dtype (a dtype label here
            # invites misreading the poll latency as a numeric benchmark)
            "dtype": None,
            "data": "synthetic" if data_source == "mock" else "live-telemetry",
            "config": {
                "model": "gpud-amd accelerator poll cycle",
                "mode": "single-process" if args.single_process else "rank-per-gpu",
                "components_per_cycle": len(accel_components),
                "gpus_per_rank": n_gpus_seen
                if args.single_process
                else (1 if data_source == "amdsmi" else n_gpus_seen),
                "data_source": data_source,
                "poll_cycle_p99_ms": round(p99, 4),
                "daemon_cpu_percent": round(cpu_pct_max, 2),
                # deployed overhead: cycles run once per 60 s poll interval
                "projected_cpu_percent_at_60s_interval": round(
                    100.0 * (mean_ms / 1000.0) / 60.0, 5
                ),
                "mfma_bf16_tflops_per_gpu": mfma_tflops,
            },
        }
        if args.fault_replay:
            out["config"]["fault_replay"] = {
                "injected_cycles": replay["injected_cycles"],
                "detected_cycles": replay["detected"],
            }
        print(json.dumps(out))

    core.close()
    if distributed:
        dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
