"""egrpc client: blocking gRPC-over-HTTP/2 unary + server-stream calls on a
unix socket. One connection per Channel; calls are serialized (the kubelet
contracts are strictly request/response per plugin socket). Streaming calls
should use their own Channel (documented; the agent does).
"""
from __future__ import annotations

import socket
import struct
import threading
import time
from typing import Iterator, List, Optional, Tuple

from . import core, hpack
from .core import ConnectionClosed, EgrpcError, frame_header, parse_frame_header

try:
    from elastic_gpu_agent_amd import _etransport
except ImportError:  # pure-Python fallback (also forced by EGPU_PY_TRANSPORT=1)
    _etransport = None


def _use_native() -> bool:
    import os

    return _etransport is not None and os.environ.get("EGPU_PY_TRANSPORT") != "1"


class Channel:
    def __init__(self, unix_path: str, connect_timeout: float = 10.0,
                 authority: bytes = b"localhost", _force_python: bool = False):
        self._path = unix_path
        self._authority = authority
        self._lock = threading.RLock()
        self._sock: Optional[socket.socket] = None
        self._buf = bytearray()
        self._decoder: Optional[hpack.Decoder] = None
        self._next_stream = 1
        self.peer_max_frame = core.DEFAULT_MAX_FRAME
        self.peer_initial_window = core.DEFAULT_WINDOW
        self.conn_send_window = core.DEFAULT_WINDOW
        self._connect_timeout = connect_timeout
        self._conn_recv_deficit = 0
        self._stream_recv_deficit = 0
        self._deficit_sid = 0
        self._active_stream_window = core.DEFAULT_WINDOW
        # C++ unary data plane (native/etransport.cpp); streaming calls
        # delegate to a lazily-created pure-Python sibling channel
        self._native = None
        self._stream_channel: Optional["Channel"] = None
        if not _force_python and _use_native():
            self._native = _etransport.ClientCore(unix_path)

    # ---- connection ----
    def _connect(self) -> None:
        deadline = time.time() + self._connect_timeout
        last_err: Optional[Exception] = None
        while time.time() < deadline:
            try:
                sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
                sock.connect(self._path)
                break
            except OSError as e:
                last_err = e
                time.sleep(0.05)
        else:
            raise EgrpcError(core.UNAVAILABLE, f"connect {self._path}: {last_err}")
        self._sock = sock
        self._buf = bytearray()
        self._decoder = hpack.Decoder()
        self._next_stream = 1
        self.conn_send_window = core.DEFAULT_WINDOW
        self.peer_initial_window = core.DEFAULT_WINDOW
        self.peer_max_frame = core.DEFAULT_MAX_FRAME
        sock.sendall(
            core.PREFACE
            + core.settings_frame([
                (core.SETTINGS_MAX_FRAME_SIZE, core.OUR_MAX_FRAME),
                (core.SETTINGS_INITIAL_WINDOW_SIZE, core.RECV_WINDOW),
            ])
            + core.window_update(0, core.RECV_WINDOW - core.DEFAULT_WINDOW)
        )

    def _ensure(self) -> None:
        if self._sock is None:
            self._connect()

    def close(self) -> None:
        with self._lock:
            if self._native is not None:
                self._native.close()
            if self._stream_channel is not None:
                self._stream_channel.close()
                self._stream_channel = None
            if self._sock is not None:
                try:
                    self._sock.close()
                except OSError:
                    pass
                self._sock = None

    def _reset(self) -> None:
        if self._sock is not None:
            try:
                self._sock.close()
            except OSError:
                pass
        self._sock = None

    # ---- frame IO ----
    def _read_frame(self):
        hdr = core.read_exact(self._sock, 9, self._buf)
        length, ftype, flags, sid = parse_frame_header(hdr)
        body = core.read_exact(self._sock, length, self._buf) if length else b""
        return ftype, flags, sid, body

    def _handle_conn_frame(self, ftype, flags, sid, body) -> bool:
        """Process connection-level frames; returns True if consumed."""
        if ftype == core.SETTINGS:
            if not flags & core.FLAG_ACK:
                for off in range(0, len(body), 6):
                    k, v = struct.unpack_from(">HI", body, off)
                    if k == core.SETTINGS_MAX_FRAME_SIZE:
                        self.peer_max_frame = v
                    elif k == core.SETTINGS_INITIAL_WINDOW_SIZE:
                        # RFC 7540 §6.9.2: the delta applies to every open
                        # stream's window, including one mid-send
                        self._active_stream_window += v - self.peer_initial_window
                        self.peer_initial_window = v
                self._sock.sendall(core.settings_frame([], flags=core.FLAG_ACK))
            return True
        if ftype == core.PING:
            if not flags & core.FLAG_ACK:
                self._sock.sendall(frame_header(8, core.PING, core.FLAG_ACK, 0) + body)
            return True
        if ftype == core.WINDOW_UPDATE and sid == 0:
            (inc,) = struct.unpack(">I", body)
            self.conn_send_window += inc
            return True
        if ftype == core.GOAWAY:
            raise ConnectionClosed()
        return False

    def _replenish(self, sid: int, consumed: int, stream_done: bool) -> None:
        """Lazy flow-control replenishment (we advertise RECV_WINDOW; batch
        updates at half-consumption to avoid per-frame syscalls)."""
        self._conn_recv_deficit += consumed
        self._stream_recv_deficit = self._stream_recv_deficit + consumed \
            if sid == self._deficit_sid else consumed
        self._deficit_sid = sid
        upd = b""
        if self._conn_recv_deficit >= core.RECV_WINDOW // 2:
            upd += core.window_update(0, self._conn_recv_deficit)
            self._conn_recv_deficit = 0
        if self._stream_recv_deficit >= core.RECV_WINDOW // 2 and not stream_done:
            upd += core.window_update(sid, self._stream_recv_deficit)
            self._stream_recv_deficit = 0
        if upd:
            self._sock.sendall(upd)

    def _request_headers(self, path: bytes) -> bytes:
        return hpack.encode_headers([
            (b":method", b"POST"),
            (b":scheme", b"http"),
            (b":path", path),
            (b":authority", self._authority),
            (b"content-type", b"application/grpc"),
            (b"te", b"trailers"),
        ])

    def _send_request(self, path: bytes, message: bytes,
                      block: Optional[bytes] = None) -> int:
        sid = self._next_stream
        self._next_stream += 2
        if block is None:
            block = self._request_headers(path)
        payload = core.grpc_frame(message)
        self._active_stream_window = self.peer_initial_window
        out = bytearray(frame_header(len(block), core.HEADERS, core.FLAG_END_HEADERS, sid))
        out += block
        # send DATA respecting windows; pump frames when blocked
        off, total = 0, len(payload)
        view = memoryview(payload)
        while True:
            avail = min(self.conn_send_window, self._active_stream_window,
                        self.peer_max_frame)
            if total - off > 0 and avail <= 0:
                self._sock.sendall(out)
                out = bytearray()
                ftype, flags, fsid, body = self._read_frame()
                if not self._handle_conn_frame(ftype, flags, fsid, body):
                    if ftype == core.WINDOW_UPDATE and fsid == sid:
                        (inc,) = struct.unpack(">I", body)
                        self._active_stream_window += inc
                    elif ftype == core.RST_STREAM:
                        raise EgrpcError(core.UNAVAILABLE, "stream reset during send")
                continue
            n = min(total - off, avail)
            last = off + n >= total
            flags = core.FLAG_END_STREAM if last else 0
            out += frame_header(n, core.DATA, flags, sid) + bytes(view[off : off + n])
            self.conn_send_window -= n
            self._active_stream_window -= n
            off += n
            if last:
                break
        self._sock.sendall(out)
        return sid

    def _read_response(self, sid: int):
        """Returns (messages, trailers_map). Raises EgrpcError on bad status."""
        messages: List[bytes] = []
        data = bytearray()
        headers: List[Tuple[bytes, bytes]] = []
        while True:
            ftype, flags, fsid, body = self._read_frame()
            if self._handle_conn_frame(ftype, flags, fsid, body):
                continue
            if fsid != sid:
                continue  # stale stream traffic
            if ftype in (core.HEADERS, core.CONTINUATION):
                off = 0
                pad = 0
                if ftype == core.HEADERS and flags & core.FLAG_PADDED:
                    pad = body[0]
                    off = 1
                if ftype == core.HEADERS and flags & core.FLAG_PRIORITY:
                    off += 5
                headers += self._decoder.decode(body[off : len(body) - pad])
                if flags & core.FLAG_END_STREAM:
                    hmap = {n: v for n, v in headers}
                    status = int(hmap.get(b"grpc-status", b"0"))
                    if status != 0:
                        raise EgrpcError(status,
                                         core.percent_decode(hmap.get(b"grpc-message", b"")))
                    messages.extend(core.parse_grpc_frames(bytes(data)))
                    return messages, hmap
            elif ftype == core.DATA:
                data += body
                if len(body):
                    self._replenish(sid, len(body), stream_done=False)
                if flags & core.FLAG_END_STREAM:  # no trailers (non-gRPC peer)
                    raise EgrpcError(core.INTERNAL, "stream ended without trailers")
            elif ftype == core.RST_STREAM:
                raise EgrpcError(core.UNAVAILABLE, "stream reset")

    # ---- public API ----
    def _native_connect(self):
        deadline = time.time() + self._connect_timeout
        last = None
        while time.time() < deadline:
            try:
                self._native.connect()
                return
            except RuntimeError as e:
                last = e
                time.sleep(0.05)
        raise EgrpcError(core.UNAVAILABLE, f"connect {self._path}: {last}")

    def unary_unary(self, path: str, request_serializer=None, response_deserializer=None):
        pbytes = path.encode()
        if self._native is not None:
            header_block = self._request_headers(pbytes)
            native = self._native

            def native_call(request, timeout: Optional[float] = None):
                msg = request_serializer(request) if request_serializer else request
                with self._lock:
                    if not native.connected():
                        self._native_connect()
                    try:
                        status, data, gmsg = native.call_unary(
                            header_block, msg, timeout or 0.0
                        )
                    except RuntimeError as e:
                        native.close()
                        if "timeout" in str(e):
                            raise EgrpcError(core.DEADLINE_EXCEEDED, path) from e
                        raise EgrpcError(core.UNAVAILABLE,
                                         f"connection lost: {e}") from e
                if status != 0:
                    raise EgrpcError(status, gmsg)
                return response_deserializer(data) if response_deserializer else data

            return native_call
        # the header block is constant per method (stateless encoder):
        # precompute it once instead of per call
        header_block = self._request_headers(pbytes)

        def call(request, timeout: Optional[float] = None):
            msg = request_serializer(request) if request_serializer else request
            with self._lock:
                self._ensure()
                prev_to = self._sock.gettimeout()
                self._sock.settimeout(timeout)
                try:
                    sid = self._send_request(pbytes, msg, header_block)
                    msgs, _ = self._read_response(sid)
                except socket.timeout as e:
                    self._reset()
                    raise EgrpcError(core.DEADLINE_EXCEEDED, path) from e
                except (ConnectionClosed, OSError) as e:
                    self._reset()
                    raise EgrpcError(core.UNAVAILABLE, f"connection lost: {e}") from e
                finally:
                    if self._sock is not None:
                        self._sock.settimeout(prev_to)
            raw = msgs[0] if msgs else b""
            return response_deserializer(raw) if response_deserializer else raw

        return call

    def unary_stream(self, path: str, request_serializer=None, response_deserializer=None):
        if self._native is not None:
            if self._stream_channel is None:
                self._stream_channel = Channel(
                    self._path, self._connect_timeout, self._authority,
                    _force_python=True,
                )
            return self._stream_channel.unary_stream(
                path, request_serializer, response_deserializer
            )
        pbytes = path.encode()

        def call(request, timeout: Optional[float] = None) -> Iterator:
            msg = request_serializer(request) if request_serializer else request
            with self._lock:
                self._ensure()
                self._sock.settimeout(timeout)
                sid = self._send_request(pbytes, msg)

                def gen():
                    pending = bytearray()
                    with self._lock:
                        while True:
                            ftype, flags, fsid, body = self._read_frame()
                            if self._handle_conn_frame(ftype, flags, fsid, body):
                                continue
                            if fsid != sid:
                                continue
                            if ftype == core.DATA:
                                pending += body
                                if len(body):
                                    self._replenish(sid, len(body), stream_done=False)
                                while len(pending) >= 5:
                                    (mlen,) = struct.unpack_from(">I", pending, 1)
                                    if len(pending) < 5 + mlen:
                                        break
                                    raw = bytes(pending[5 : 5 + mlen])
                                    del pending[: 5 + mlen]
                                    yield (response_deserializer(raw)
                                           if response_deserializer else raw)
                                if flags & core.FLAG_END_STREAM:
                                    return
                            elif ftype in (core.HEADERS, core.CONTINUATION):
                                hdrs = self._decoder.decode(body)
                                if flags & core.FLAG_END_STREAM:
                                    hmap = {n: v for n, v in hdrs}
                                    status = int(hmap.get(b"grpc-status", b"0"))
                                    if status != 0:
                                        raise EgrpcError(
                                            status,
                                            core.percent_decode(
                                                hmap.get(b"grpc-message", b"")))
                                    return
                            elif ftype == core.RST_STREAM:
                                raise EgrpcError(core.UNAVAILABLE, "stream reset")

                return gen()

        return call
