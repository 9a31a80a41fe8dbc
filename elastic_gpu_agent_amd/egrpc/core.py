"""Shared HTTP/2 + gRPC framing primitives for the egrpc transport."""
from __future__ import annotations

import struct
from typing import Tuple

PREFACE = b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n"

# frame types
DATA = 0x0
HEADERS = 0x1
PRIORITY = 0x2
RST_STREAM = 0x3
SETTINGS = 0x4
PUSH_PROMISE = 0x5
PING = 0x6
GOAWAY = 0x7
WINDOW_UPDATE = 0x8
CONTINUATION = 0x9

# flags
FLAG_END_STREAM = 0x1
FLAG_ACK = 0x1
FLAG_END_HEADERS = 0x4
FLAG_PADDED = 0x8
FLAG_PRIORITY = 0x20

# settings ids
SETTINGS_HEADER_TABLE_SIZE = 0x1
SETTINGS_ENABLE_PUSH = 0x2
SETTINGS_MAX_CONCURRENT_STREAMS = 0x3
SETTINGS_INITIAL_WINDOW_SIZE = 0x4
SETTINGS_MAX_FRAME_SIZE = 0x5
SETTINGS_MAX_HEADER_LIST_SIZE = 0x6

DEFAULT_WINDOW = 65535
DEFAULT_MAX_FRAME = 16384

# how much receive window we grant peers (big, to keep large ListAndWatch /
# Allocate payloads from stalling on flow control)
RECV_WINDOW = 32 * 1024 * 1024
OUR_MAX_FRAME = 1 * 1024 * 1024

# gRPC status codes (subset used by the agent)
OK = 0
CANCELLED = 1
UNKNOWN = 2
INVALID_ARGUMENT = 3
DEADLINE_EXCEEDED = 4
NOT_FOUND = 5
RESOURCE_EXHAUSTED = 8
FAILED_PRECONDITION = 9
INTERNAL = 13
UNAVAILABLE = 14

_HDR = struct.Struct(">I")  # helper for 4-byte ints


def frame_header(length: int, ftype: int, flags: int, stream_id: int) -> bytes:
    return length.to_bytes(3, "big") + bytes((ftype, flags)) + _HDR.pack(stream_id)


def parse_frame_header(buf: bytes) -> Tuple[int, int, int, int]:
    length = int.from_bytes(buf[:3], "big")
    return length, buf[3], buf[4], _HDR.unpack_from(buf, 5)[0] & 0x7FFFFFFF


def settings_frame(pairs, flags: int = 0) -> bytes:
    body = b"".join(struct.pack(">HI", k, v) for k, v in pairs)
    return frame_header(len(body), SETTINGS, flags, 0) + body


def window_update(stream_id: int, increment: int) -> bytes:
    return frame_header(4, WINDOW_UPDATE, 0, stream_id) + _HDR.pack(increment)


def rst_stream(stream_id: int, code: int) -> bytes:
    return frame_header(4, RST_STREAM, 0, stream_id) + _HDR.pack(code)


def goaway(last_stream: int, code: int) -> bytes:
    return frame_header(8, GOAWAY, 0, 0) + _HDR.pack(last_stream) + _HDR.pack(code)


def grpc_frame(message: bytes) -> bytes:
    """gRPC length-prefixed message (uncompressed)."""
    return b"\x00" + _HDR.pack(len(message)) + message


def parse_grpc_frames(body: bytes):
    out = []
    pos = 0
    while pos + 5 <= len(body):
        compressed = body[pos]
        (length,) = _HDR.unpack_from(body, pos + 1)
        if compressed:
            raise EgrpcError(INTERNAL, "compressed gRPC frames not supported")
        out.append(body[pos + 5 : pos + 5 + length])
        pos += 5 + length
    return out


def percent_encode(msg: str) -> bytes:
    out = bytearray()
    for b in msg.encode("utf-8"):
        if 0x20 <= b <= 0x7E and b != 0x25:
            out.append(b)
        else:
            out += b"%%%02X" % b
    return bytes(out)


def percent_decode(raw: bytes) -> str:
    out = bytearray()
    i = 0
    while i < len(raw):
        if raw[i] == 0x25 and i + 2 < len(raw):
            try:
                out.append(int(raw[i + 1 : i + 3], 16))
                i += 3
                continue
            except ValueError:
                pass
        out.append(raw[i])
        i += 1
    return out.decode("utf-8", "replace")


class EgrpcError(Exception):
    """gRPC-status-carrying error (both transport ends raise/serialize it)."""

    def __init__(self, code: int, message: str = ""):
        super().__init__(f"grpc-status {code}: {message}")
        self._code = code
        self._message = message

    def code(self) -> int:
        return self._code

    def details(self) -> str:
        return self._message


class ConnectionClosed(Exception):
    pass


def read_exact(sock, n: int, buf: bytearray) -> bytes:
    """Read exactly n bytes using buf as the carry-over buffer."""
    while len(buf) < n:
        chunk = sock.recv(262144)
        if not chunk:
            raise ConnectionClosed()
        buf += chunk
    out = bytes(buf[:n])
    del buf[:n]
    return out
