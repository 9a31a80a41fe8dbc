"""ListAndWatch behavior: encoded stream ≡ dict stream, refresh triggers,
and device-change re-advertisement."""
import threading
import time

import pytest

from elastic_gpu_agent_amd.protos import deviceplugin as dp

from helpers import Harness


@pytest.fixture
def h(tmp_path):
    harness = Harness(str(tmp_path), gpus=2)
    harness.plugin.cfg.options.health_refresh_seconds = 0.2
    yield harness
    harness.close()


def test_encoded_stream_matches_dict_stream(h):
    enc_gen = h.plugin.core.list_and_watch_encoded(None)
    payload = next(enc_gen)
    decoded = dp.ListAndWatchResponse.decode(payload)
    dict_gen = h.plugin.core.list_and_watch(None)
    direct = next(dict_gen)
    assert decoded == direct
    enc_gen.close()
    dict_gen.close()


class _Ctx:
    def __init__(self):
        self.active = True

    def is_active(self):
        return self.active


def test_readvertise_on_device_change(h):
    """A GPU disappearing from enumeration shrinks the advertised list."""
    ctx = _Ctx()
    gen = h.plugin.core.list_and_watch(ctx)
    first = next(gen)
    assert len(first["devices"]) == 200

    got = []

    def consume():
        for resp in gen:
            got.append(resp)
            break

    t = threading.Thread(target=consume, daemon=True)
    t.start()
    # GPU 1 falls off the bus
    h.operator.backend.count = 1
    deadline = time.time() + 5
    while not got and time.time() < deadline:
        time.sleep(0.05)
    assert got, "no re-advertisement after enumeration change"
    assert len(got[0]["devices"]) == 100
    ctx.active = False
    gen.close()


def test_trigger_refresh_resends(h):
    ctx = _Ctx()
    gen = h.plugin.core.list_and_watch(ctx)
    next(gen)
    got = []

    def consume():
        for resp in gen:
            got.append(resp)
            break

    t = threading.Thread(target=consume, daemon=True)
    t.start()
    time.sleep(0.05)
    h.plugin.core.trigger_refresh()
    deadline = time.time() + 5
    while not got and time.time() < deadline:
        time.sleep(0.05)
    assert got and len(got[0]["devices"]) == 200
    ctx.active = False
    gen.close()


def test_mib_unit_snapshot_through_grpcio(tmp_path):
    """294,912-device ListAndWatch snapshot (1 GPU at 1-MiB units, ~7 MB on
    the wire) streamed by the native server and decoded by the real gRPC
    stack — the scale the default memory-unit contract implies."""
    import grpc

    from helpers import Harness

    h = Harness(str(tmp_path), gpus=1, mem_unit_mib=1)
    try:
        h.plugin.memory_server.serve()
        h.plugin.memory_server.wait_ready()
        ch = grpc.insecure_channel(
            f"unix://{h.plugin.memory_server.socket_path}",
            options=[("grpc.max_receive_message_length", 64 << 20)],
        )
        from elastic_gpu_agent_amd.protos import deviceplugin as dp

        stream = ch.unary_stream(
            dp.METHOD_LIST_AND_WATCH,
            request_serializer=dp.Empty.encode,
            response_deserializer=dp.ListAndWatchResponse.decode,
        )({})
        first = next(stream)
        assert len(first["devices"]) == 288 * 1024  # 294,912
        assert first["devices"][0]["ID"] == "0-000000"
        assert first["devices"][-1]["ID"] == f"0-{288 * 1024 - 1:06d}"
        stream.cancel()
        ch.close()
    finally:
        h.close()


def test_unhealthy_after_repeated_enumeration_failure(h):
    """3 consecutive enumeration failures → everything re-advertised
    Unhealthy; recovery → Healthy again."""
    ctx = _Ctx()
    gen = h.plugin.core.list_and_watch(ctx)
    first = next(gen)
    assert all(d["health"] == "Healthy" for d in first["devices"])

    real_devices = h.operator.backend.devices

    def boom():
        raise RuntimeError("amdsmi wedged")

    got = []

    def consume(n):
        for resp in gen:
            got.append(resp)
            if len(got) >= n:
                break

    h.operator.backend.devices = boom
    t = threading.Thread(target=lambda: consume(1), daemon=True)
    t.start()
    deadline = time.time() + 10
    while not got and time.time() < deadline:
        time.sleep(0.05)
    assert got, "no Unhealthy re-advertisement"
    assert all(d["health"] == "Unhealthy" for d in got[0]["devices"])

    # recovery
    h.operator.backend.devices = real_devices
    t2 = threading.Thread(target=lambda: consume(2), daemon=True)
    t2.start()
    deadline = time.time() + 10
    while len(got) < 2 and time.time() < deadline:
        time.sleep(0.05)
    assert len(got) >= 2, "no recovery re-advertisement"
    assert all(d["health"] == "Healthy" for d in got[1]["devices"])
    ctx.active = False
    gen.close()
