"""Domain-type tests: device-set hashing and pod state (de)serialization.

The hash and JSON shapes must stay byte-compatible with the reference's state
(ref: pkg/types/device.go:49-54 — sha256 of ":".join(sorted ids), first 8 hex
chars; pkg/types/pod.go:51-62 — JSON container→Device map keyed ns/name).
"""
import hashlib

from elastic_gpu_agent_amd.types import Device, GPUDevice, PodContainer, PodInfo


def test_device_hash_matches_reference_scheme():
    ids = ["1-03", "0-17", "0-02"]
    d = Device.new(ids, "elasticgpu.io/gpu-core")
    expected = hashlib.sha256(b"0-02:0-17:1-03").hexdigest()[:8]
    assert d.hash == expected
    assert list(d.list) == ["0-02", "0-17", "1-03"]


def test_device_hash_known_value():
    # pinned vector so accidental hash-scheme changes fail loudly
    d = Device.new(["a", "b"], "")
    assert d.hash == hashlib.sha256(b"a:b").hexdigest()[:8]
    assert len(d.hash) == 8


def test_device_equals():
    a = Device.new(["x", "y"], "r")
    b = Device.new(["y", "x"], "r")
    c = Device.new(["y", "x"], "other")
    assert a.equals(b)
    assert not a.equals(c)


def test_podinfo_roundtrip_go_field_names():
    pi = PodInfo(namespace="ns", name="pod")
    pi.container_device_map["main"] = Device.new(["0-00", "0-01"], "elasticgpu.io/gpu-core")
    raw = pi.val()
    assert b'"Hash"' in raw and b'"List"' in raw and b'"ResourceName"' in raw
    back = PodInfo.from_raw(pi.key(), raw)
    assert back.namespace == "ns" and back.name == "pod"
    assert back.container_device_map["main"].equals(pi.container_device_map["main"])


def test_podinfo_bad_key():
    import pytest

    with pytest.raises(ValueError):
        PodInfo.from_raw("no-slash", b"{}")


def test_pod_container_strings():
    pc = PodContainer("ns", "pod", "c1")
    assert str(pc) == "ns/pod:c1"
    assert pc.pod() == "ns/pod"


def test_gpu_device_memory_mib():
    g = GPUDevice(uuid="u", index=0, memory_bytes=288 * 1024**3)
    assert g.memory_mib == 288 * 1024
