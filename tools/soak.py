"""Soak driver: sustained concurrent pod churn against a live agent stack.

Runs N worker threads binding/unbinding fractional pods over the real unix
sockets while a ListAndWatch stream stays open and the GC loop runs, for
--seconds. Reports RPC counts/errors and RSS growth; exits non-zero on any
error or RSS growth beyond --rss-limit-mb.

Usage:  python tools/soak.py --seconds 60 --workers 4
        (gpurun: real amdsmi backend is picked up automatically on a GPU box)
"""
from __future__ import annotations

import argparse
import os
import sys
import tempfile
import threading
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))


def rss_mb() -> float:
    with open("/proc/self/status") as f:
        for line in f:
            if line.startswith("VmRSS"):
                return int(line.split()[1]) / 1024.0
    return 0.0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=int, default=60)
    ap.add_argument("--workers", type=int, default=4)
    ap.add_argument("--rss-limit-mb", type=float, default=150.0,
                    help="fixed RSS growth allowance (with the GC-cadence "
                    "malloc_trim, RSS plateaus after the initial ramp)")
    ap.add_argument("--rss-per-cycle-bytes", type=float, default=50.0,
                    help="additional allowance per bind cycle")
    args = ap.parse_args()

    from helpers import Harness, PluginClient

    from elastic_gpu_agent_amd import consts
    from elastic_gpu_agent_amd.types import Device, PodContainer

    tmp = tempfile.mkdtemp(prefix="egpu-soak-")
    h = Harness(tmp, gpus=2)
    try:
        import torch

        if torch.cuda.is_available():
            from elastic_gpu_agent_amd.isolation import CUMaskAllocator
            from elastic_gpu_agent_amd.operator import GPUOperator
            from elastic_gpu_agent_amd.operator.amdsmi import AmdSmiBackend

            backend = AmdSmiBackend()
            h.plugin.cfg.operator = GPUOperator(backend, dev_root=h.paths.dev_root)
            h.plugin.cfg.cumask = CUMaskAllocator(h.storage, backend.devices())
            print(f"# soak on real backend: {len(backend.devices())} GPU(s)")
    except Exception:
        pass
    gpus = h.plugin.cfg.operator.devices()

    h.plugin.core_server.serve()
    h.plugin.memory_server.serve()
    h.plugin.core_server.wait_ready()
    h.plugin.memory_server.wait_ready()

    stop = threading.Event()
    errors = []
    counts = {"alloc": 0, "prestart": 0, "gc": 0, "law_msgs": 0}
    lock = threading.Lock()

    def worker(wid: int):
        client_core = PluginClient(h.plugin.core_server.socket_path)
        client_mem = PluginClient(h.plugin.memory_server.socket_path)
        i = 0
        while not stop.is_set():
            i += 1
            gpu = gpus[(wid + i) % len(gpus)].index
            name = f"soak-{wid}-{i}"
            try:
                if i % 2 == 0:
                    # each worker owns a disjoint 25-slot range per GPU: real
                    # kubelet never assigns one device ID to two live pods,
                    # so concurrent pods must not share ID sets (= hashes)
                    pct = 5 + (i % 4) * 5
                    base = (wid % 4) * 25
                    start = base + (i % (25 - 20))
                    ids = [f"{gpu}-{(start + k):02d}" for k in range(pct)]
                    d = Device.new(ids, consts.RESOURCE_GPU_CORE)
                    h.core_locator.assign(d.hash, PodContainer("soak", name, "c"))
                    h.add_assumed_pod("soak", name, "c", str(gpu))
                    client_core.allocate({"container_requests": [{"devicesIDs": ids}]})
                    with lock:
                        counts["alloc"] += 1
                    client_core.pre_start({"devicesIDs": ids})
                else:
                    units = 4 + i % 8
                    base = wid * 300 + (i * 17) % 250  # worker-unique: no cross-worker hash reuse
                    ids = [f"{gpu}-{base + k:06d}" for k in range(units)]
                    d = Device.new(ids, consts.RESOURCE_GPU_MEMORY)
                    h.mem_locator.assign(d.hash, PodContainer("soak", name, "c"))
                    h.add_assumed_pod("soak", name, "c", str(gpu))
                    client_mem.allocate({"container_requests": [{"devicesIDs": ids}]})
                    with lock:
                        counts["alloc"] += 1
                    client_mem.pre_start({"devicesIDs": ids})
                with lock:
                    counts["prestart"] += 1
                h.sitter.remove("soak", name)
                # harness hygiene: each worker cleans its own locator entry
                # (kubelet's bookkeeping stand-in) so RSS reflects the agent
                locator = h.core_locator if i % 2 == 0 else h.mem_locator
                locator.table.pop(d.hash, None)
            except Exception as e:
                errors.append(f"worker {wid} iter {i}: {e!r}")
                return
        client_core.close()
        client_mem.close()

    def watcher():
        client = PluginClient(h.plugin.core_server.socket_path)
        try:
            stream = client.list_and_watch({})
            for _msg in stream:
                with lock:
                    counts["law_msgs"] += 1
                if stop.is_set():
                    return
        except Exception as e:
            if not stop.is_set():
                errors.append(f"watcher: {e!r}")
        finally:
            client.close()

    def gc_loop():
        from elastic_gpu_agent_amd.plugins.aggregate import _malloc_trim

        while not stop.is_set():
            time.sleep(1.0)
            try:
                h.plugin.gc_once()
                _malloc_trim()  # production GC cadence does the same
                with lock:
                    counts["gc"] += 1
            except Exception as e:
                errors.append(f"gc: {e!r}")
                return

    rss0 = rss_mb()
    threads = [threading.Thread(target=worker, args=(w,), daemon=True)
               for w in range(args.workers)]
    threads += [threading.Thread(target=watcher, daemon=True),
                threading.Thread(target=gc_loop, daemon=True)]
    for t in threads:
        t.start()
    t_end = time.time() + args.seconds
    while time.time() < t_end and not errors:
        time.sleep(1.0)
    stop.set()
    time.sleep(1.5)
    h.plugin.core.trigger_refresh()  # unblock the watcher stream

    # drain the full backlog, return freed pages, then measure
    from elastic_gpu_agent_amd.plugins.aggregate import _malloc_trim

    while h.plugin.gc_once() > 0:
        pass
    _malloc_trim()
    rss1 = rss_mb()
    h.close()
    growth = rss1 - rss0
    allowed = args.rss_limit_mb + counts["prestart"] * args.rss_per_cycle_bytes / 1e6
    print(f"soak: {counts} rss {rss0:.1f} -> {rss1:.1f} MB "
          f"(+{growth:.1f}, allowed {allowed:.1f})")
    if errors:
        print("ERRORS:")
        for e in errors[:20]:
            print(" ", e)
        return 1
    if growth > allowed:
        print(f"RSS growth {growth:.1f} MB exceeds allowance {allowed:.1f}")
        return 1
    print("soak OK")
    return 0


if __name__ == "__main__":
    sys.exit(main())
