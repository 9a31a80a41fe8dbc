"""Randomized pod-churn stress: after every GC pass the node state must be
internally consistent — symlinks exist exactly for live bound pods, storage
mirrors the live set, masks are immutable once assigned, and a new mask
overlaps live ones ONLY when the GPU genuinely lacks free CU pairs (pair
rounding counts: 5% = 7 pairs, not 6.4)."""
import os
import random

from elastic_gpu_agent_amd import consts
from elastic_gpu_agent_amd.isolation.cumask import parse_mask_hex
from elastic_gpu_agent_amd.types import Device, PodContainer

from helpers import Harness


def test_random_churn_invariants(tmp_path):
    rng = random.Random(1234)
    h = Harness(str(tmp_path), gpus=2)
    live = {}  # name -> (Device, gpu_index, percent, bound_under_capacity, coresidents)
    counter = 0

    def bind_pod():
        nonlocal counter
        counter += 1
        name = f"pod-{counter}"
        gpu = rng.randrange(2)
        percent = rng.choice([5, 10, 25, 40])
        start = rng.randrange(100 - percent)
        ids = [f"{gpu}-{(start + i):02d}" for i in range(percent)]
        d = Device.new(ids, consts.RESOURCE_GPU_CORE)
        h.core_locator.assign(d.hash, PodContainer("ns", name, "main"))
        h.add_assumed_pod("ns", name, "main", str(gpu))
        h.plugin.core.allocate({"container_requests": [{"devicesIDs": ids}]}, None)
        h.plugin.core.pre_start_container({"devicesIDs": ids}, None)
        # the precise no-steal invariant, checked at bind time: if the GPU
        # still had enough FREE pairs, the new mask must not overlap any
        # live mask
        from elastic_gpu_agent_amd.isolation.cumask import cu_count_for_percent

        # union over the ALLOCATOR's view: every recorded mask on this GPU
        # (including killed-but-not-yet-GC'd pods — their processes may still
        # be draining, so the allocator rightly treats their CUs as busy)
        import json as _json

        new_mask = parse_mask_hex(h.plugin.cfg.cumask.get(d.hash)["cu_mask"])
        union = [0] * len(new_mask)
        for key, val in h.storage.aux_items("mask/"):
            if key == "mask/" + d.hash:
                continue
            rec = _json.loads(val)
            if rec.get("gpu_index") != gpu:
                continue
            m2 = parse_mask_hex(rec["cu_mask"])
            union = [a | b for a, b in zip(union, m2)]
        overlap = any(a & b for a, b in zip(new_mask, union))
        want_pairs = (cu_count_for_percent(percent) + 1) // 2
        free_pairs = 128 - sum(bin(w).count("1") for w in union) // 2
        if overlap:
            assert free_pairs < want_pairs, (
                f"{name} overlaps although {free_pairs} pairs were free "
                f"(wanted {want_pairs})"
            )
        live[name] = (d, gpu, percent, new_mask)

    def kill_pod():
        name = rng.choice(list(live))
        h.sitter.remove("ns", name)
        del live[name]

    def check_invariants():
        # 1. symlinks exist exactly for live pods
        links = {f for f in os.listdir(h.paths.dev_root) if f.startswith("elastic-gpu-")
                 and not f.startswith("elastic-gpuctl-")}
        expected = {f"elastic-gpu-{v[0].hash}-0" for v in live.values()}
        assert links == expected, (links, expected)
        # 2. storage mirrors live set
        stored = []
        h.storage.for_each(lambda pi: stored.append(pi.name))
        assert sorted(stored) == sorted(live)
        # 3. masks are immutable once assigned (pods keep the CUs they got)
        for name, (d, g, pct, mask_at_bind) in live.items():
            rec = h.plugin.cfg.cumask.get(d.hash)
            assert rec is not None, f"mask record lost for live {name}"
            assert parse_mask_hex(rec["cu_mask"]) == mask_at_bind, (
                f"mask of {name} changed after bind"
            )

    for step in range(120):
        if not live or (len(live) < 8 and rng.random() < 0.6):
            bind_pod()
        else:
            kill_pod()
        if rng.random() < 0.5 or step % 10 == 0:
            h.plugin.gc_once()
            check_invariants()
    # drain
    for name in list(live):
        h.sitter.remove("ns", name)
    live.clear()
    h.plugin.gc_once()
    check_invariants()
    h.close()
