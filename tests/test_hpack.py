"""HPACK codec tests.

The Huffman table (written from RFC 7541 Appendix B) is machine-validated
against libnghttp2's HPACK inflater when available: our encoder's output for
every byte value must decode to the original through nghttp2. Interop with
grpcio's (C-core) encoder is covered separately in test_egrpc.py over live
connections.
"""
import ctypes
import ctypes.util
import os
import random

import pytest

from elastic_gpu_agent_amd.egrpc import hpack


def test_integer_roundtrip():
    for prefix in (4, 5, 6, 7):
        for v in (0, 1, (1 << prefix) - 2, (1 << prefix) - 1, 127, 128, 300, 2**20):
            buf = hpack.encode_int(v, prefix)
            out, pos = hpack.decode_int(buf, 0, prefix)
            assert out == v and pos == len(buf)


def test_huffman_roundtrip_all_bytes():
    data = bytes(range(256)) * 3 + b"grpc-status" + b"/v1beta1.DevicePlugin/Allocate"
    assert hpack.huffman_decode(hpack.huffman_encode(data)) == data


def test_huffman_roundtrip_random():
    rng = random.Random(7)
    for _ in range(200):
        data = bytes(rng.randrange(256) for _ in range(rng.randrange(1, 64)))
        assert hpack.huffman_decode(hpack.huffman_encode(data)) == data


def test_encode_decode_static_and_literal():
    headers = [
        (b":method", b"POST"),
        (b":scheme", b"http"),
        (b":path", b"/v1beta1.DevicePlugin/Allocate"),
        (b":authority", b"localhost"),
        (b"content-type", b"application/grpc"),
        (b"te", b"trailers"),
        (b"x-custom", b"some value"),
    ]
    buf = hpack.encode_headers(headers)
    dec = hpack.Decoder()
    assert dec.decode(buf) == headers


def test_decoder_dynamic_table():
    # simulate a peer that adds entries with incremental indexing and then
    # references them
    dec = hpack.Decoder()
    block1 = bytes([0x40]) + hpack.encode_int(8, 7) + b"x-header" + \
        hpack.encode_int(5, 7) + b"one!!"
    assert dec.decode(block1) == [(b"x-header", b"one!!")]
    # index 62 = first dynamic entry
    block2 = hpack.encode_int(62, 7, 0x80)
    assert dec.decode(block2) == [(b"x-header", b"one!!")]
    # table size update to 0 evicts
    block3 = hpack.encode_int(0, 5, 0x20)
    dec.decode(block3)
    with pytest.raises(hpack.HpackError):
        dec.decode(block2)


# ---- nghttp2 cross-validation ----------------------------------------------

NGHTTP2 = "/opt/conda/lib/libnghttp2.so"


class _NV(ctypes.Structure):
    # pointers as void* (c_char_p would stop at embedded NULs)
    _fields_ = [
        ("name", ctypes.c_void_p),
        ("value", ctypes.c_void_p),
        ("namelen", ctypes.c_size_t),
        ("valuelen", ctypes.c_size_t),
        ("flags", ctypes.c_uint8),
    ]


@pytest.mark.skipif(not os.path.exists(NGHTTP2), reason="libnghttp2 not present")
def test_huffman_table_against_nghttp2_inflater():
    """Our huffman_encode for EVERY symbol must inflate correctly through
    nghttp2 — proves the Appendix-B table is transcribed right."""
    lib = ctypes.CDLL(NGHTTP2)
    lib.nghttp2_hd_inflate_new.argtypes = [ctypes.POINTER(ctypes.c_void_p)]
    lib.nghttp2_hd_inflate_hd2.restype = ctypes.c_ssize_t
    lib.nghttp2_hd_inflate_hd2.argtypes = [
        ctypes.c_void_p, ctypes.POINTER(_NV), ctypes.POINTER(ctypes.c_int),
        ctypes.POINTER(ctypes.c_uint8), ctypes.c_size_t, ctypes.c_int,
    ]

    def inflate_literal(name: bytes, value: bytes):
        """Build a literal-without-indexing header with OUR huffman coding of
        name and value, inflate via nghttp2, return (name, value)."""
        hn = hpack.huffman_encode(name)
        hv = hpack.huffman_encode(value)
        block = (b"\x00" + hpack.encode_int(len(hn), 7, 0x80) + hn
                 + hpack.encode_int(len(hv), 7, 0x80) + hv)
        inflater = ctypes.c_void_p()
        assert lib.nghttp2_hd_inflate_new(ctypes.byref(inflater)) == 0
        buf = (ctypes.c_uint8 * len(block)).from_buffer_copy(block)
        nv = _NV()
        flags = ctypes.c_int(0)
        rv = lib.nghttp2_hd_inflate_hd2(
            inflater, ctypes.byref(nv), ctypes.byref(flags), buf, len(block), 1
        )
        assert rv > 0, f"nghttp2 inflate failed rv={rv}"
        out = (
            ctypes.string_at(nv.name, nv.namelen),
            ctypes.string_at(nv.value, nv.valuelen),
        )
        lib.nghttp2_hd_inflate_del(inflater)
        return out

    # every byte value appears in some value string
    for lo in range(0, 256, 32):
        value = bytes(range(lo, lo + 32))
        name = b"x-test"
        n, v = inflate_literal(name, value)
        assert n == name
        assert v == value, f"huffman mismatch in byte range {lo}-{lo+31}"

    # realistic strings
    for s in (b"/v1beta1.DevicePlugin/Allocate", b"application/grpc",
              b"grpc-go/1.27.0", b"trailers", b"0", b"elasticgpu.io/gpu-core"):
        n, v = inflate_literal(b"p", s)
        assert v == s


@pytest.mark.skipif(not os.path.exists(NGHTTP2), reason="libnghttp2 not present")
def test_differential_fuzz_against_nghttp2_deflater():
    """Differential fuzz: random header lists encoded by nghttp2's HPACK
    deflater (which uses indexed entries, dynamic-table references, Huffman
    and table-size updates like real kubelet/grpc encoders) must decode
    identically through our connection-scoped decoder — including dynamic
    table state carried ACROSS blocks on one connection."""
    lib = ctypes.CDLL(NGHTTP2)
    lib.nghttp2_hd_deflate_new.argtypes = [ctypes.POINTER(ctypes.c_void_p), ctypes.c_size_t]
    lib.nghttp2_hd_deflate_hd.restype = ctypes.c_ssize_t
    lib.nghttp2_hd_deflate_hd.argtypes = [
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint8), ctypes.c_size_t,
        ctypes.POINTER(_NV), ctypes.c_size_t,
    ]

    rng = random.Random(20260913)
    names = [b":path", b":method", b":authority", b"content-type", b"te",
             b"grpc-timeout", b"user-agent", b"x-custom-header", b"x-trace-id",
             b"grpc-encoding", b"authorization"]
    values = [b"POST", b"/v1beta1.DevicePlugin/Allocate", b"application/grpc",
              b"trailers", b"10S", b"grpc-go/1.27.1 (linux; amd64)", b"",
              b"elasticgpu.io/gpu-core", b"0", b"identity"]

    def random_headers():
        out = []
        for _ in range(rng.randrange(1, 10)):
            if rng.random() < 0.7:
                n = rng.choice(names)
            else:
                n = bytes(rng.choice(b"abcdefghijklmnopqrstuvwxyz-")
                          for _ in range(rng.randrange(1, 20)))
            if rng.random() < 0.6:
                v = rng.choice(values)
            else:
                v = bytes(rng.randrange(0x20, 0x7F)
                          for _ in range(rng.randrange(0, 40)))
            out.append((n, v))
        return out

    for conn_round in range(30):  # 30 fresh "connections"
        deflater = ctypes.c_void_p()
        assert lib.nghttp2_hd_deflate_new(ctypes.byref(deflater), 4096) == 0
        dec = hpack.Decoder()
        for block_round in range(10):  # 10 header blocks per connection
            headers = random_headers()
            nva = (_NV * len(headers))()
            keepalive = []
            for i, (n, v) in enumerate(headers):
                nb, vb = ctypes.create_string_buffer(n, len(n)), \
                    ctypes.create_string_buffer(v, len(v))
                keepalive += [nb, vb]
                nva[i].name = ctypes.cast(nb, ctypes.c_void_p)
                nva[i].value = ctypes.cast(vb, ctypes.c_void_p)
                nva[i].namelen = len(n)
                nva[i].valuelen = len(v)
                nva[i].flags = 0
            buf = (ctypes.c_uint8 * 65536)()
            rv = lib.nghttp2_hd_deflate_hd(deflater, buf, 65536, nva, len(headers))
            assert rv > 0, f"deflate failed rv={rv}"
            block = bytes(buf[:rv])
            decoded = dec.decode(block)
            assert decoded == headers, (
                f"conn {conn_round} block {block_round}: {decoded} != {headers}"
            )
        lib.nghttp2_hd_deflate_del(deflater)


@pytest.mark.skipif(not os.path.exists(NGHTTP2), reason="libnghttp2 not present")
def test_cpp_decoder_differential_fuzz():
    """The C++ transport core's HPACK decoder (native/etransport.cpp) under
    the same nghttp2-deflater fuzz as the Python decoder."""
    try:
        from elastic_gpu_agent_amd import _etransport
    except ImportError:
        pytest.skip("_etransport not built")
    lib = ctypes.CDLL(NGHTTP2)
    lib.nghttp2_hd_deflate_new.argtypes = [ctypes.POINTER(ctypes.c_void_p), ctypes.c_size_t]
    lib.nghttp2_hd_deflate_hd.restype = ctypes.c_ssize_t
    lib.nghttp2_hd_deflate_hd.argtypes = [
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint8), ctypes.c_size_t,
        ctypes.POINTER(_NV), ctypes.c_size_t,
    ]
    rng = random.Random(777)
    for conn_round in range(20):
        deflater = ctypes.c_void_p()
        assert lib.nghttp2_hd_deflate_new(ctypes.byref(deflater), 4096) == 0
        dec = _etransport.HpackTester()
        for block_round in range(10):
            headers = [
                (bytes(rng.choice(b"abcdefgh-:") for _ in range(rng.randrange(1, 12))),
                 bytes(rng.randrange(0x20, 0x7F) for _ in range(rng.randrange(0, 30))))
                for _ in range(rng.randrange(1, 8))
            ]
            # mix in realistic repeats that exercise dynamic-table indexing
            headers += [(b":path", b"/v1beta1.DevicePlugin/Allocate"),
                        (b"content-type", b"application/grpc")]
            nva = (_NV * len(headers))()
            keepalive = []
            for i, (n, v) in enumerate(headers):
                nb = ctypes.create_string_buffer(n, len(n))
                vb = ctypes.create_string_buffer(v, len(v))
                keepalive += [nb, vb]
                nva[i].name = ctypes.cast(nb, ctypes.c_void_p)
                nva[i].value = ctypes.cast(vb, ctypes.c_void_p)
                nva[i].namelen = len(n)
                nva[i].valuelen = len(v)
                nva[i].flags = 0
            buf = (ctypes.c_uint8 * 65536)()
            rv = lib.nghttp2_hd_deflate_hd(deflater, buf, 65536, nva, len(headers))
            assert rv > 0
            decoded = dec.decode(bytes(buf[:rv]))
            assert decoded == headers, f"conn {conn_round} block {block_round}"
        lib.nghttp2_hd_deflate_del(deflater)
