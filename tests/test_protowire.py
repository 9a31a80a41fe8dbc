"""Wire-codec tests: roundtrips of the kubelet proto messages and
cross-checks against the `protobuf` runtime (present in the image) to prove
our hand-rolled encoding is wire-compatible."""
import pytest

from elastic_gpu_agent_amd.protos import deviceplugin as dp
from elastic_gpu_agent_amd.protos import podresources as pr
from elastic_gpu_agent_amd.protos.protowire import decode_varint, encode_varint


@pytest.mark.parametrize("n", [0, 1, 127, 128, 300, 2**21, 2**63 - 1])
def test_varint_roundtrip(n):
    buf = encode_varint(n)
    val, pos = decode_varint(buf, 0)
    assert val == n and pos == len(buf)


def test_varint_negative():
    buf = encode_varint(-1)
    assert len(buf) == 10
    val, _ = decode_varint(buf, 0)
    assert val == 2**64 - 1


def test_register_request_roundtrip():
    msg = {
        "version": "v1beta1",
        "endpoint": "elastic-gpushare-core.sock",
        "resource_name": "elasticgpu.io/gpu-core",
        "options": {"pre_start_required": True, "get_preferred_allocation_available": True},
    }
    buf = dp.RegisterRequest.encode(msg)
    out = dp.RegisterRequest.decode(buf)
    assert out["version"] == "v1beta1"
    assert out["resource_name"] == "elasticgpu.io/gpu-core"
    assert out["options"]["pre_start_required"] is True
    assert out["options"]["get_preferred_allocation_available"] is True


def test_list_and_watch_roundtrip():
    msg = {
        "devices": [
            {"ID": f"0-{i:02d}", "health": "Healthy", "topology": {"nodes": [{"ID": 3}]}}
            for i in range(100)
        ]
    }
    out = dp.ListAndWatchResponse.decode(dp.ListAndWatchResponse.encode(msg))
    assert len(out["devices"]) == 100
    assert out["devices"][7]["ID"] == "0-07"
    assert out["devices"][7]["topology"]["nodes"][0]["ID"] == 3


def test_allocate_roundtrip_with_maps():
    msg = {
        "container_responses": [
            {
                "envs": {"GPU": "abcd1234", "HIP_VISIBLE_DEVICES": "0"},
                "mounts": [{"container_path": "/a", "host_path": "/b", "read_only": True}],
                "devices": [
                    {"container_path": "/dev/kfd", "host_path": "/dev/kfd", "permissions": "rwm"}
                ],
                "annotations": {"k": "v"},
            }
        ]
    }
    out = dp.AllocateResponse.decode(dp.AllocateResponse.encode(msg))
    cr = out["container_responses"][0]
    assert cr["envs"]["GPU"] == "abcd1234"
    assert cr["mounts"][0]["read_only"] is True
    assert cr["devices"][0]["host_path"] == "/dev/kfd"
    assert cr["annotations"] == {"k": "v"}


def test_empty_messages():
    assert dp.Empty.encode({}) == b""
    assert dp.Empty.decode(b"") == {}
    assert dp.PreStartContainerResponse.decode(b"") == {}


def test_unknown_fields_skipped():
    # a future-proto peer may send unknown fields; they must be skipped
    from elastic_gpu_agent_amd.protos.protowire import encode_varint as ev

    extra = ev(99 << 3 | 0) + ev(7)  # unknown varint field 99
    buf = dp.PreStartContainerRequest.encode({"devicesIDs": ["a", "b"]}) + extra
    out = dp.PreStartContainerRequest.decode(buf)
    assert out["devicesIDs"] == ["a", "b"]


def test_podresources_roundtrip():
    msg = {
        "pod_resources": [
            {
                "name": "pod-1",
                "namespace": "default",
                "containers": [
                    {
                        "name": "main",
                        "devices": [
                            {"resource_name": "elasticgpu.io/gpu-core", "device_ids": ["0-00"]}
                        ],
                    }
                ],
            }
        ]
    }
    out = pr.ListPodResourcesResponse.decode(pr.ListPodResourcesResponse.encode(msg))
    pod = out["pod_resources"][0]
    assert pod["namespace"] == "default"
    assert pod["containers"][0]["devices"][0]["device_ids"] == ["0-00"]


def test_cross_check_against_protobuf_runtime():
    """Encode with our codec, decode with google.protobuf (and vice versa)."""
    from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

    pool = descriptor_pool.DescriptorPool()
    fd = descriptor_pb2.FileDescriptorProto()
    fd.name = "xcheck.proto"
    fd.package = "xcheck"
    fd.syntax = "proto3"
    m = fd.message_type.add()
    m.name = "RegisterRequest"
    for i, fname in [(1, "version"), (2, "endpoint"), (3, "resource_name")]:
        f = m.field.add()
        f.name = fname
        f.number = i
        f.type = descriptor_pb2.FieldDescriptorProto.TYPE_STRING
        f.label = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
    pool.Add(fd)
    cls = message_factory.GetMessageClass(pool.FindMessageTypeByName("xcheck.RegisterRequest"))

    ours = dp.RegisterRequest.encode(
        {"version": "v1beta1", "endpoint": "e.sock", "resource_name": "r"}
    )
    theirs = cls.FromString(ours)
    assert theirs.version == "v1beta1"
    assert theirs.endpoint == "e.sock"
    assert theirs.resource_name == "r"

    pb = cls(version="v2", endpoint="x", resource_name="y").SerializeToString()
    out = dp.RegisterRequest.decode(pb)
    assert out["version"] == "v2" and out["endpoint"] == "x" and out["resource_name"] == "y"
