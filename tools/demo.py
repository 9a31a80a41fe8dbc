"""End-to-end demo: the full fractional-GPU story on one node, narrated.

Runs against the real MI355X when present (amdsmi backend + masked workload
verification through the HSA shim), or the fake gfx950 fleet on CPU:

  1. enumerate GPUs (libamd_smi)
  2. schedule three pods with the scheduler-sim (GetPreferredAllocation RPC)
  3. kubelet flow per pod: Allocate → podresources record → PreStartContainer
  4. inspect: symlinks, CU masks, limits files, persisted state
  5. [GPU] run the census kernel inside each pod's mask → distinct-CU counts
  6. occupancy report, then delete pods → GC reclaims everything

Usage: python tools/demo.py            (transcript to stdout)
"""
from __future__ import annotations

import json
import os
import subprocess
import sys
import tempfile

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))


def log(msg):
    print(msg, flush=True)


def main():
    from helpers import Harness, PluginClient

    from elastic_gpu_agent_amd import consts
    from elastic_gpu_agent_amd.schedsim import SimScheduler
    from elastic_gpu_agent_amd.types import Device, PodContainer

    on_gpu = False
    try:
        import torch

        on_gpu = torch.cuda.is_available()
    except Exception:
        pass

    tmp = tempfile.mkdtemp(prefix="egpu-demo-")
    h = Harness(tmp, gpus=2)
    if on_gpu:
        from elastic_gpu_agent_amd.isolation import CUMaskAllocator
        from elastic_gpu_agent_amd.operator import GPUOperator
        from elastic_gpu_agent_amd.operator.amdsmi import AmdSmiBackend

        backend = AmdSmiBackend()
        h.plugin.cfg.operator = GPUOperator(backend, dev_root=h.paths.dev_root)
        h.plugin.cfg.cumask = CUMaskAllocator(h.storage, backend.devices())

    gpus = h.plugin.cfg.operator.devices()
    log(f"== 1. enumeration ({'libamd_smi' if on_gpu else 'fake gfx950 fleet'}) ==")
    for g in gpus:
        log(f"   gpu{g.index}: {g.uuid}  render=/dev/dri/renderD{g.drm_render_minor}  "
            f"{g.cu_count} CUs / {g.xcd_count} XCDs  {g.memory_bytes >> 30} GiB HBM3E  "
            f"{g.compute_partition}  numa={g.numa_node}")

    h.plugin.core_server.serve()
    h.plugin.memory_server.serve()
    h.plugin.core_server.wait_ready()
    h.plugin.memory_server.wait_ready()
    core = PluginClient(h.plugin.core_server.socket_path)
    mem = PluginClient(h.plugin.memory_server.socket_path)

    def preferred(resource, avail, size):
        resp = core.preferred({"container_requests": [{
            "available_deviceIDs": avail, "must_include_deviceIDs": [],
            "allocation_size": size}]})
        return resp["container_responses"][0]["deviceIDs"]

    sched = SimScheduler(gpus, mem_unit_mib=1024)
    requests = [("pod-a", 25, 0), ("pod-b", 50, 0), ("pod-c", 0, 72)]  # %, GiB
    log("== 2. scheduling (sim of elastic-gpu-scheduler; xGMI/NUMA-aware) ==")
    placements = {}
    for name, pct, gib in requests:
        p = sched.place("main", core_units=pct, memory_units=gib,
                        preferred_fn=preferred if pct else None)
        assert p, f"no capacity for {name}"
        placements[name] = (p, pct, gib)
        log(f"   {name}: core={pct}% mem={gib} GiB -> gpu{p['gpu_indexes']} "
            f"annotations={p['annotations']}")

    log("== 3. kubelet flow: Allocate -> PreStartContainer per pod ==")
    bound = {}
    for name, (p, pct, gib) in placements.items():
        ids = p["core_ids"] or p["memory_ids"]
        res = consts.RESOURCE_GPU_CORE if pct else consts.RESOURCE_GPU_MEMORY
        d = Device.new(ids, res)
        locator = h.core_locator if pct else h.mem_locator
        client = core if pct else mem
        locator.assign(d.hash, PodContainer("demo", name, "main"))
        from elastic_gpu_agent_amd.kube.pods import Pod

        h.sitter.add(Pod(namespace="demo", name=name, annotations=p["annotations"]))
        resp = client.allocate({"container_requests": [{"devicesIDs": ids}]})
        cr = resp["container_responses"][0]
        client.pre_start({"devicesIDs": ids})
        bound[name] = d
        log(f"   {name}: GPU={cr['envs']['GPU']}  "
            f"shim={'HSA_TOOLS_LIB' in cr['envs']}  devices="
            f"{[s['host_path'] for s in cr.get('devices', [])]}")

    log("== 4. node state after binding ==")
    for name, d in bound.items():
        limits = h.plugin.cfg.limits.read(d.hash)
        link = os.path.join(h.paths.dev_root, f"elastic-gpu-{d.hash}-0")
        log(f"   {name}: link {os.path.basename(link)} -> {os.readlink(link)}")
        log(f"          limits: {json.dumps(limits)}")

    if on_gpu:
        log("== 5. isolation verified on the GPU (census kernel per pod mask) ==")
        shim = os.path.join(REPO, "elastic_gpu_agent_amd", "libegpu_shim.so")
        for name, d in bound.items():
            limits = h.plugin.cfg.limits.read(d.hash)
            if "cu_mask" not in limits:
                continue
            env = dict(os.environ)
            env.update({"HSA_TOOLS_LIB": shim, "EGPU_CU_MASK": limits["cu_mask"]})
            out = subprocess.run(
                [sys.executable, "-c",
                 "from elastic_gpu_agent_amd.isolation import probes; "
                 "print(len(probes.census(0, 2048, 100000)))"],
                env=env, cwd=REPO, capture_output=True, text=True, timeout=300)
            seen = int(out.stdout.strip().splitlines()[-1])
            log(f"   {name}: mask allows {limits['cu_count']} CUs -> census saw "
                f"{seen} distinct CUs  "
                f"{'OK' if seen <= limits['cu_count'] else 'VIOLATION'}")
            assert seen <= limits["cu_count"]

    log("== 6. occupancy / teardown ==")
    from elastic_gpu_agent_amd.isolation.occupancy import report

    class _NoSmi:
        def gpu_utilization(self, idx):
            raise RuntimeError("n/a on CPU")

        def gpu_processes(self, idx):
            return []

    smi = None if on_gpu else _NoSmi()
    if on_gpu:
        from elastic_gpu_agent_amd import _amdsmi as smi  # noqa: F811
    rep = report(h.storage, h.plugin.cfg.limits, state_dir=tmp, smi=smi)
    for row in rep["pods"]:
        log(f"   {row['pod']}: {row['resource']}={row['units']}u "
            f"gpu={row['gpu_index']} cu_limit={row['cu_limit']} "
            f"mem_limit={row['mem_limit_bytes']}")
    for name in bound:
        h.sitter.remove("demo", name)
    reclaimed = h.plugin.gc_once()
    log(f"   deleted pods -> GC reclaimed {reclaimed} records; dev links left: "
        f"{[f for f in os.listdir(h.paths.dev_root) if f.startswith('elastic-gpu-')]}")
    core.close()
    mem.close()
    h.close()
    log("demo OK")
    return 0


if __name__ == "__main__":
    sys.exit(main())
