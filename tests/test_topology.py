"""xGMI-aware preferred-allocation tests."""
from elastic_gpu_agent_amd.operator.fake import FakeBackend
from elastic_gpu_agent_amd.topology import gpu_index_of, group_by_gpu, prefer_allocation, xgmi_score


def devs(n=8):
    return FakeBackend(count=n).devices()


def test_gpu_index_parsing():
    assert gpu_index_of("3-07") == 3
    assert gpu_index_of("12-000042") == 12
    groups = group_by_gpu(["0-00", "0-01", "1-00"])
    assert set(groups) == {0, 1} and len(groups[0]) == 2


def test_prefer_packs_single_gpu():
    # 30 units available on GPU0 (partially used) and 100 on GPU1:
    # a 20-unit ask should pack onto the fuller GPU0, keeping GPU1 whole.
    avail = [f"0-{i:02d}" for i in range(30)] + [f"1-{i:02d}" for i in range(100)]
    picked = prefer_allocation(avail, [], 20, devs(2))
    assert len(picked) == 20
    assert all(d.startswith("0-") for d in picked)


def test_prefer_honors_must_include():
    avail = [f"{g}-{i:02d}" for g in range(2) for i in range(100)]
    must = ["1-00", "1-01"]
    picked = prefer_allocation(avail, must, 10, devs(2))
    assert set(must) <= set(picked)
    # fill continues on the GPU already touched by must_include
    assert all(d.startswith("1-") for d in picked)


def test_prefer_multi_gpu_spills():
    avail = [f"0-{i:02d}" for i in range(100)] + [f"1-{i:02d}" for i in range(100)]
    picked = prefer_allocation(avail, [], 150, devs(2))
    assert len(picked) == 150
    assert sum(1 for d in picked if d.startswith("0-")) == 100


def test_xgmi_score_full_mesh():
    d = {g.index: g for g in devs(8)}
    assert xgmi_score([0, 1, 2, 3], d) == 6  # all pairs linked on a full mesh
    assert xgmi_score([0], d) == 0


def test_single_gpu_never_spans():
    """Fractional requests must come from ONE GPU even when packing would
    prefer spreading over fragmented ones (a spanning fractional set is
    unbindable at PreStart — found via the drain/schedsim test)."""
    # GPU 0 has only 20 free, GPU 1 has 40 free
    avail = [f"0-{i:02d}" for i in range(20)] + [f"1-{i:02d}" for i in range(40)]
    picked = prefer_allocation(avail, [], 30, devs(2), single_gpu=True)
    assert len(picked) == 30
    assert all(d.startswith("1-") for d in picked)


def test_single_gpu_short_pick_when_nothing_fits():
    avail = [f"0-{i:02d}" for i in range(20)] + [f"1-{i:02d}" for i in range(25)]
    picked = prefer_allocation(avail, [], 30, devs(2), single_gpu=True)
    assert len(picked) < 30  # caller treats as does-not-fit


def test_single_gpu_follows_must_include():
    avail = [f"{g}-{i:02d}" for g in range(2) for i in range(50)]
    picked = prefer_allocation(avail, ["1-07"], 10, devs(2), single_gpu=True)
    assert len(picked) == 10
    assert all(d.startswith("1-") for d in picked)
