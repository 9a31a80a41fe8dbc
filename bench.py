"""Flagship benchmark: fractional-pod churn against the device-plugin stack.

Measures the BASELINE.json metric — "fractional pods/GPU + p50 Allocate() RPC
latency at 1/2/4/8 MI355X" — by driving the full agent pipeline over real
gRPC unix sockets, one agent instance (one rank) per GPU:

  step = for each of --pods-per-gpu pods on this rank's GPU:
           Allocate(gpu-core) + Allocate(gpu-memory)   [the timed RPCs]
           PreStartContainer(core) + PreStartContainer(memory)
         then delete all pods and run one GC reconciliation pass.

The kubelet side is simulated in-process (gRPC client over UDS + podresources
assignment table + pod annotations), the GPU side is real when available:
on a GPU box enumeration goes through libamd_smi and PreStart materializes
symlinks to the node's actual /dev/dri/renderD* minors.

Output: ONE JSON line on rank 0 (driver contract), value = whole-job pod
allocation-cycle throughput (cycles/s summed over ranks), plus the p50/p99
Allocate RPC latency in µs.

Configs (--config): mixed (default; core fractions + memory fractions, 16
pods/GPU oversubscribed), whole-gpu, mem-fraction, compute-fraction,
kind-fake (forces the fake backend).
"""
from __future__ import annotations

import argparse
import json
import os
import statistics
import sys
import tempfile
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))


def get_dist():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    return rank, world


def pod_plan(config: str, pods_per_gpu: int, gpu_index: int, mem_total_mib: int,
             mem_unit_mib: int):
    """Returns a list of (kind, ids) per pod for one GPU."""
    plans = []
    if config == "whole-gpu":
        plans.append(("core", [f"{gpu_index}-{i:02d}" for i in range(100)]))
        return plans
    if config == "mem-fraction":
        # 4 pods × 1/4 of HBM (the 72 GiB config on 288 GB parts)
        units = mem_total_mib // mem_unit_mib // 4
        for p in range(4):
            plans.append(
                ("mem", [f"{gpu_index}-{p * units + i:06d}" for i in range(units)])
            )
        return plans
    if config == "compute-fraction":
        # 8 pods × 12% of the card (CU-mask partitions)
        for p in range(8):
            plans.append(("core", [f"{gpu_index}-{(p * 12 + i) % 100:02d}" for i in range(12)]))
        return plans
    # mixed (default): pods_per_gpu pods alternating core fractions and
    # memory fractions, oversubscribed core (sum > 100%)
    for p in range(pods_per_gpu):
        if p % 2 == 0:
            frac = 10 + (p % 5) * 5
            start = (p * 7) % 90
            plans.append(
                ("core", [f"{gpu_index}-{(start + i) % 100:02d}" for i in range(frac)])
            )
        else:
            units = max(1, mem_total_mib // mem_unit_mib // max(pods_per_gpu, 1))
            base = (p // 2) * units
            plans.append(("mem", [f"{gpu_index}-{base + i:06d}" for i in range(units)]))
    return plans


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--pods-per-gpu", type=int, default=16)
    ap.add_argument("--config", default="mixed",
                    choices=["mixed", "whole-gpu", "mem-fraction", "compute-fraction",
                             "kind-fake"])
    ap.add_argument("--mem-unit-mib", type=int, default=1024)
    args = ap.parse_args()

    rank, world = get_dist()
    dist = None
    if world > 1:
        import torch.distributed as tdist

        tdist.init_process_group(backend="gloo")
        dist = tdist

    import torch

    from elastic_gpu_agent_amd import consts
    from elastic_gpu_agent_amd.metrics import GLOBAL_METRICS
    from elastic_gpu_agent_amd.types import Device, PodContainer
    from helpers import Harness, PluginClient

    # ---- backend selection: real amdsmi on a GPU box, fake otherwise ----
    backend = None
    backend_name = "fake-gfx950"
    if args.config != "kind-fake" and torch.cuda.is_available():
        try:
            from elastic_gpu_agent_amd.operator.amdsmi import AmdSmiBackend

            backend = AmdSmiBackend()
            backend.devices()
            backend_name = "amdsmi"
        except Exception as e:
            print(f"# amdsmi backend unavailable ({e}); using fake", file=sys.stderr)
            backend = None

    tmp = tempfile.mkdtemp(prefix=f"egpu-bench-r{rank}-")
    h = Harness(tmp, gpus=1, mem_unit_mib=args.mem_unit_mib)
    if backend is not None:
        from elastic_gpu_agent_amd.isolation import CUMaskAllocator
        from elastic_gpu_agent_amd.operator import GPUOperator

        h.plugin.cfg.operator = GPUOperator(backend, dev_root=h.paths.dev_root)
        h.plugin.cfg.cumask = CUMaskAllocator(h.storage, backend.devices())
    gpus = h.plugin.cfg.operator.devices()
    gpu = gpus[min(rank, len(gpus) - 1)]
    mem_total_mib = gpu.memory_mib

    # serve both resource plugins over real unix sockets
    h.plugin.core_server.serve()
    h.plugin.memory_server.serve()
    h.plugin.core_server.wait_ready()
    h.plugin.memory_server.wait_ready()
    core = PluginClient(h.plugin.core_server.socket_path)
    mem = PluginClient(h.plugin.memory_server.socket_path)

    plans = pod_plan(args.config, args.pods_per_gpu, gpu.index, mem_total_mib,
                     args.mem_unit_mib)
    pods_per_step = len(plans)

    alloc_lat = []  # seconds, every Allocate RPC
    prestart_lat = []  # seconds, every PreStartContainer RPC

    def one_step(step_i: int):
        pods = []
        for p, (kind, ids) in enumerate(plans):
            ns, name, container = "bench", f"pod-{step_i}-{p}", "main"
            res = consts.RESOURCE_GPU_CORE if kind == "core" else consts.RESOURCE_GPU_MEMORY
            d = Device.new(ids, res)
            locator = h.core_locator if kind == "core" else h.mem_locator
            locator.assign(d.hash, PodContainer(ns, name, container))
            h.add_assumed_pod(ns, name, container, str(gpu.index))
            client = core if kind == "core" else mem
            t0 = time.perf_counter()
            client.allocate({"container_requests": [{"devicesIDs": ids}]})
            t1 = time.perf_counter()
            client.pre_start({"devicesIDs": ids})
            t2 = time.perf_counter()
            alloc_lat.append(t1 - t0)
            prestart_lat.append(t2 - t1)
            pods.append((ns, name, d))
        # teardown: pods deleted, GC reclaims symlinks/masks/limits/state
        for ns, name, _ in pods:
            h.sitter.remove(ns, name)
        reclaimed = h.plugin.gc_once()
        assert reclaimed == len(pods), f"GC reclaimed {reclaimed}/{len(pods)}"

    def barrier():
        if dist is not None:
            dist.barrier()
        if torch.cuda.is_available():
            torch.cuda.synchronize()

    for w in range(args.warmup):
        one_step(-1 - w)

    alloc_lat.clear()
    prestart_lat.clear()
    barrier()
    t_start = time.perf_counter()
    for s in range(args.steps):
        one_step(s)
    barrier()
    elapsed = time.perf_counter() - t_start

    # max elapsed over ranks (slowest rank defines the job); worst-rank p50/p99
    lat_sorted = sorted(alloc_lat)
    p50_us = lat_sorted[len(lat_sorted) // 2] * 1e6
    p99_us = lat_sorted[min(len(lat_sorted) - 1, int(0.99 * len(lat_sorted)))] * 1e6
    ps_sorted = sorted(prestart_lat)
    p50_prestart_us = ps_sorted[len(ps_sorted) // 2] * 1e6 if ps_sorted else 0.0
    elapsed_max = elapsed
    if dist is not None:
        t = torch.tensor([elapsed, p50_us, p99_us])
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed_max, p50_us, p99_us = (float(x) for x in t)

    cycles_total = world * pods_per_step * args.steps
    value = cycles_total / elapsed_max

    core.close()
    mem.close()
    h.close()

    if rank == 0:
        print(json.dumps({
            "metric": "fractional pods/GPU + p50 Allocate() RPC latency at 1/2/4/8 MI355X",
            "value": round(value, 2),
            "unit": "pod-allocation-cycles/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed_max / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "n/a",
            "data": "synthetic",
            "p50_allocate_us": round(p50_us, 1),
            "p99_allocate_us": round(p99_us, 1),
            "p50_prestart_us": round(p50_prestart_us, 1),
            "config": {
                "model": "gpushare-device-plugin",
                "scenario": args.config,
                "pods_per_gpu": pods_per_step,
                "backend": backend_name,
                "mem_unit_mib": args.mem_unit_mib,
                "global_batch": pods_per_step * world,
                "seq_len": 0,
                "parallelism": f"dp{world} (1 agent per GPU)",
            },
        }))

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
