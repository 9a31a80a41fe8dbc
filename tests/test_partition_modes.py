"""CU-mask allocation under non-SPX compute partitions: in CPX each agent is
one XCD (32 CUs); masks must size and spread against the partition's shape,
not the physical card's."""
from elastic_gpu_agent_amd import consts
from elastic_gpu_agent_amd.isolation import CUMaskAllocator
from elastic_gpu_agent_amd.isolation.cumask import parse_mask_hex, popcount
from elastic_gpu_agent_amd.storage import Storage
from elastic_gpu_agent_amd.types import GPUDevice


def cpx_device(index=0):
    return GPUDevice(
        uuid=f"GPU-cpx-{index}", index=index, memory_bytes=36 * 1024**3,
        drm_render_minor=128 + index, cu_count=32, xcd_count=1,
        compute_partition="CPX",
    )


def test_mask_allocation_on_cpx_slice(tmp_db):
    st = Storage(tmp_db)
    alloc = CUMaskAllocator(st, [cpx_device()])
    hexmask, n = alloc.allocate("h1", 0, 50)  # 50% of a 32-CU slice
    words = parse_mask_hex(hexmask)
    assert n == 16
    assert popcount(words) == 16
    assert all(w == 0 for w in words[1:])  # only CU 0..31 addressable
    # second pod gets the other half, disjoint
    hexmask2, n2 = alloc.allocate("h2", 0, 50)
    w2 = parse_mask_hex(hexmask2)
    assert popcount(w2) == 16
    assert all((a & b) == 0 for a, b in zip(words, w2))
    st.close()


def test_mask_allocation_dpx_shape(tmp_db):
    st = Storage(tmp_db)
    dpx = GPUDevice(
        uuid="GPU-dpx", index=0, memory_bytes=144 * 1024**3,
        drm_render_minor=128, cu_count=128, xcd_count=4,
        compute_partition="DPX",
    )
    alloc = CUMaskAllocator(st, [dpx])
    hexmask, n = alloc.allocate("h1", 0, 25)
    words = parse_mask_hex(hexmask)
    assert n == 32
    per_xcd = [0, 0, 0, 0]
    for w_i, w in enumerate(words):
        for b in range(32):
            if w >> b & 1:
                per_xcd[(w_i * 32 + b) // 32] += 1
    assert per_xcd == [8, 8, 8, 8]  # even across the partition's 4 XCDs
    st.close()
