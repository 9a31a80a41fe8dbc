"""HPACK (RFC 7541) — the header codec for the egrpc transport.

Decoder: complete — static + dynamic table, all literal forms, table-size
updates, Huffman decoding (table from RFC 7541 Appendix B; validated in the
test-suite against libnghttp2's deflater for every byte value and against
grpcio's encoder over live connections).

Encoder: deliberately minimal — static-table indexing where an exact match
exists, literal-without-indexing otherwise, never Huffman, never dynamic
entries. That is always legal HPACK, keeps the encoder stateless, and lets
hot-path header blocks be precomputed once as raw bytes.
"""
from __future__ import annotations

from typing import List, Tuple

# ---- RFC 7541 Appendix A: static table (1-based) ----------------------------
STATIC_TABLE: List[Tuple[bytes, bytes]] = [
    (b":authority", b""),
    (b":method", b"GET"),
    (b":method", b"POST"),
    (b":path", b"/"),
    (b":path", b"/index.html"),
    (b":scheme", b"http"),
    (b":scheme", b"https"),
    (b":status", b"200"),
    (b":status", b"204"),
    (b":status", b"206"),
    (b":status", b"304"),
    (b":status", b"400"),
    (b":status", b"404"),
    (b":status", b"500"),
    (b"accept-charset", b""),
    (b"accept-encoding", b"gzip, deflate"),
    (b"accept-language", b""),
    (b"accept-ranges", b""),
    (b"accept", b""),
    (b"access-control-allow-origin", b""),
    (b"age", b""),
    (b"allow", b""),
    (b"authorization", b""),
    (b"cache-control", b""),
    (b"content-disposition", b""),
    (b"content-encoding", b""),
    (b"content-language", b""),
    (b"content-length", b""),
    (b"content-location", b""),
    (b"content-range", b""),
    (b"content-type", b""),
    (b"cookie", b""),
    (b"date", b""),
    (b"etag", b""),
    (b"expect", b""),
    (b"expires", b""),
    (b"from", b""),
    (b"host", b""),
    (b"if-match", b""),
    (b"if-modified-since", b""),
    (b"if-none-match", b""),
    (b"if-range", b""),
    (b"if-unmodified-since", b""),
    (b"last-modified", b""),
    (b"link", b""),
    (b"location", b""),
    (b"max-forwards", b""),
    (b"proxy-authenticate", b""),
    (b"proxy-authorization", b""),
    (b"range", b""),
    (b"referer", b""),
    (b"refresh", b""),
    (b"retry-after", b""),
    (b"server", b""),
    (b"set-cookie", b""),
    (b"strict-transport-security", b""),
    (b"transfer-encoding", b""),
    (b"user-agent", b""),
    (b"vary", b""),
    (b"via", b""),
    (b"www-authenticate", b""),
]
_STATIC_EXACT = {pair: i + 1 for i, pair in enumerate(STATIC_TABLE)}
_STATIC_NAME = {}
for _i, (_n, _v) in enumerate(STATIC_TABLE):
    _STATIC_NAME.setdefault(_n, _i + 1)

# ---- RFC 7541 Appendix B: Huffman code table (symbol -> (code, bits)) -------
HUFFMAN_TABLE: List[Tuple[int, int]] = [
    (0x1FF8, 13), (0x7FFFD8, 23), (0xFFFFFE2, 28), (0xFFFFFE3, 28),
    (0xFFFFFE4, 28), (0xFFFFFE5, 28), (0xFFFFFE6, 28), (0xFFFFFE7, 28),
    (0xFFFFFE8, 28), (0xFFFFEA, 24), (0x3FFFFFFC, 30), (0xFFFFFE9, 28),
    (0xFFFFFEA, 28), (0x3FFFFFFD, 30), (0xFFFFFEB, 28), (0xFFFFFEC, 28),
    (0xFFFFFED, 28), (0xFFFFFEE, 28), (0xFFFFFEF, 28), (0xFFFFFF0, 28),
    (0xFFFFFF1, 28), (0xFFFFFF2, 28), (0x3FFFFFFE, 30), (0xFFFFFF3, 28),
    (0xFFFFFF4, 28), (0xFFFFFF5, 28), (0xFFFFFF6, 28), (0xFFFFFF7, 28),
    (0xFFFFFF8, 28), (0xFFFFFF9, 28), (0xFFFFFFA, 28), (0xFFFFFFB, 28),
    (0x14, 6), (0x3F8, 10), (0x3F9, 10), (0xFFA, 12),
    (0x1FF9, 13), (0x15, 6), (0xF8, 8), (0x7FA, 11),
    (0x3FA, 10), (0x3FB, 10), (0xF9, 8), (0x7FB, 11),
    (0xFA, 8), (0x16, 6), (0x17, 6), (0x18, 6),
    (0x0, 5), (0x1, 5), (0x2, 5), (0x19, 6),
    (0x1A, 6), (0x1B, 6), (0x1C, 6), (0x1D, 6),
    (0x1E, 6), (0x1F, 6), (0x5C, 7), (0xFB, 8),
    (0x7FFC, 15), (0x20, 6), (0xFFB, 12), (0x3FC, 10),
    (0x1FFA, 13), (0x21, 6), (0x5D, 7), (0x5E, 7),
    (0x5F, 7), (0x60, 7), (0x61, 7), (0x62, 7),
    (0x63, 7), (0x64, 7), (0x65, 7), (0x66, 7),
    (0x67, 7), (0x68, 7), (0x69, 7), (0x6A, 7),
    (0x6B, 7), (0x6C, 7), (0x6D, 7), (0x6E, 7),
    (0x6F, 7), (0x70, 7), (0x71, 7), (0x72, 7),
    (0xFC, 8), (0x73, 7), (0xFD, 8), (0x1FFB, 13),
    (0x7FFF0, 19), (0x1FFC, 13), (0x3FFC, 14), (0x22, 6),
    (0x7FFD, 15), (0x3, 5), (0x23, 6), (0x4, 5),
    (0x24, 6), (0x5, 5), (0x25, 6), (0x26, 6),
    (0x27, 6), (0x6, 5), (0x74, 7), (0x75, 7),
    (0x28, 6), (0x29, 6), (0x2A, 6), (0x7, 5),
    (0x2B, 6), (0x76, 7), (0x2C, 6), (0x8, 5),
    (0x9, 5), (0x2D, 6), (0x77, 7), (0x78, 7),
    (0x79, 7), (0x7A, 7), (0x7B, 7), (0x7FFE, 15),
    (0x7FC, 11), (0x3FFD, 14), (0x1FFD, 13), (0xFFFFFFC, 28),
    (0xFFFE6, 20), (0x3FFFD2, 22), (0xFFFE7, 20), (0xFFFE8, 20),
    (0x3FFFD3, 22), (0x3FFFD4, 22), (0x3FFFD5, 22), (0x7FFFD9, 23),
    (0x3FFFD6, 22), (0x7FFFDA, 23), (0x7FFFDB, 23), (0x7FFFDC, 23),
    (0x7FFFDD, 23), (0x7FFFDE, 23), (0xFFFFEB, 24), (0x7FFFDF, 23),
    (0xFFFFEC, 24), (0xFFFFED, 24), (0x3FFFD7, 22), (0x7FFFE0, 23),
    (0xFFFFEE, 24), (0x7FFFE1, 23), (0x7FFFE2, 23), (0x7FFFE3, 23),
    (0x7FFFE4, 23), (0x1FFFDC, 21), (0x3FFFD8, 22), (0x7FFFE5, 23),
    (0x3FFFD9, 22), (0x7FFFE6, 23), (0x7FFFE7, 23), (0xFFFFEF, 24),
    (0x3FFFDA, 22), (0x1FFFDD, 21), (0xFFFE9, 20), (0x3FFFDB, 22),
    (0x3FFFDC, 22), (0x7FFFE8, 23), (0x7FFFE9, 23), (0x1FFFDE, 21),
    (0x7FFFEA, 23), (0x3FFFDD, 22), (0x3FFFDE, 22), (0xFFFFF0, 24),
    (0x1FFFDF, 21), (0x3FFFDF, 22), (0x7FFFEB, 23), (0x7FFFEC, 23),
    (0x1FFFE0, 21), (0x1FFFE1, 21), (0x3FFFE0, 22), (0x1FFFE2, 21),
    (0x7FFFED, 23), (0x3FFFE1, 22), (0x7FFFEE, 23), (0x7FFFEF, 23),
    (0xFFFEA, 20), (0x3FFFE2, 22), (0x3FFFE3, 22), (0x3FFFE4, 22),
    (0x7FFFF0, 23), (0x3FFFE5, 22), (0x3FFFE6, 22), (0x7FFFF1, 23),
    (0x3FFFFE0, 26), (0x3FFFFE1, 26), (0xFFFEB, 20), (0x7FFF1, 19),
    (0x3FFFE7, 22), (0x7FFFF2, 23), (0x3FFFE8, 22), (0x1FFFFEC, 25),
    (0x3FFFFE2, 26), (0x3FFFFE3, 26), (0x3FFFFE4, 26), (0x7FFFFDE, 27),
    (0x7FFFFDF, 27), (0x3FFFFE5, 26), (0xFFFFF1, 24), (0x1FFFFED, 25),
    (0x7FFF2, 19), (0x1FFFE3, 21), (0x3FFFFE6, 26), (0x7FFFFE0, 27),
    (0x7FFFFE1, 27), (0x3FFFFE7, 26), (0x7FFFFE2, 27), (0xFFFFF2, 24),
    (0x1FFFE4, 21), (0x1FFFE5, 21), (0x3FFFFE8, 26), (0x3FFFFE9, 26),
    (0xFFFFFFD, 28), (0x7FFFFE3, 27), (0x7FFFFE4, 27), (0x7FFFFE5, 27),
    (0xFFFEC, 20), (0xFFFFF3, 24), (0xFFFED, 20), (0x1FFFE6, 21),
    (0x3FFFE9, 22), (0x1FFFE7, 21), (0x1FFFE8, 21), (0x7FFFF3, 23),
    (0x3FFFEA, 22), (0x3FFFEB, 22), (0x1FFFFEE, 25), (0x1FFFFEF, 25),
    (0xFFFFF4, 24), (0xFFFFF5, 24), (0x3FFFFEA, 26), (0x7FFFF4, 23),
    (0x3FFFFEB, 26), (0x7FFFFE6, 27), (0x3FFFFEC, 26), (0x3FFFFED, 26),
    (0x7FFFFE7, 27), (0x7FFFFE8, 27), (0x7FFFFE9, 27), (0x7FFFFEA, 27),
    (0x7FFFFEB, 27), (0xFFFFFFE, 28), (0x7FFFFEC, 27), (0x7FFFFED, 27),
    (0x7FFFFEE, 27), (0x7FFFFEF, 27), (0x7FFFFF0, 27), (0x3FFFFEE, 26),
    (0x3FFFFFFF, 30),  # 256 = EOS
]

# Huffman decode: a flat dict {(code, bits): symbol} would be slow; build a
# binary trie packed as lists for O(bits) walks with no allocation.
_TRIE: List[list] = [[-1, 0, 0]]  # node: [symbol, left_idx, right_idx]


def _trie_insert(code: int, bits: int, symbol: int) -> None:
    node = 0
    for i in range(bits - 1, -1, -1):
        b = (code >> i) & 1
        nxt = _TRIE[node][1 + b]
        if nxt == 0:
            _TRIE.append([-1, 0, 0])
            nxt = len(_TRIE) - 1
            _TRIE[node][1 + b] = nxt
        node = nxt
    _TRIE[node][0] = symbol


for _sym, (_code, _bits) in enumerate(HUFFMAN_TABLE):
    _trie_insert(_code, _bits, _sym)


class HpackError(ValueError):
    pass


def huffman_decode(data: bytes) -> bytes:
    out = bytearray()
    node = 0
    trie = _TRIE
    for byte in data:
        for i in range(7, -1, -1):
            node = trie[node][1 + ((byte >> i) & 1)]
            if node == 0:
                raise HpackError("invalid huffman code")
            sym = trie[node][0]
            if sym >= 0:
                if sym == 256:
                    raise HpackError("EOS in huffman data")
                out.append(sym)
                node = 0
    # trailing bits must be a prefix of EOS (all ones), <= 7 bits: they never
    # reach a symbol node, which is exactly the state we are in.
    return bytes(out)


def huffman_encode(data: bytes) -> bytes:
    """Provided for tests/completeness; the production encoder never uses it."""
    acc = 0
    nbits = 0
    out = bytearray()
    for b in data:
        code, bits = HUFFMAN_TABLE[b]
        acc = (acc << bits) | code
        nbits += bits
        while nbits >= 8:
            nbits -= 8
            out.append((acc >> nbits) & 0xFF)
    if nbits:
        out.append(((acc << (8 - nbits)) | ((1 << (8 - nbits)) - 1)) & 0xFF)
    return bytes(out)


# ---- integer primitives (RFC 7541 §5.1) -------------------------------------

def encode_int(value: int, prefix_bits: int, flags: int = 0) -> bytes:
    limit = (1 << prefix_bits) - 1
    if value < limit:
        return bytes([flags | value])
    out = bytearray([flags | limit])
    value -= limit
    while value >= 128:
        out.append(0x80 | (value & 0x7F))
        value >>= 7
    out.append(value)
    return bytes(out)


def decode_int(data: bytes, pos: int, prefix_bits: int) -> Tuple[int, int]:
    limit = (1 << prefix_bits) - 1
    value = data[pos] & limit
    pos += 1
    if value < limit:
        return value, pos
    shift = 0
    while True:
        b = data[pos]
        pos += 1
        value += (b & 0x7F) << shift
        if not b & 0x80:
            return value, pos
        shift += 7
        if shift > 56:
            raise HpackError("integer overflow")


# ---- encoder ----------------------------------------------------------------

def _encode_string(s: bytes) -> bytes:
    return encode_int(len(s), 7, 0x00) + s  # no huffman


def encode_headers(headers: List[Tuple[bytes, bytes]]) -> bytes:
    """Stateless encode: static-index exact matches, literal-without-indexing
    (0x00 prefix, static name index when available) otherwise."""
    out = bytearray()
    for name, value in headers:
        idx = _STATIC_EXACT.get((name, value))
        if idx is not None:
            out += encode_int(idx, 7, 0x80)
            continue
        nidx = _STATIC_NAME.get(name)
        if nidx is not None:
            out += encode_int(nidx, 4, 0x00)
        else:
            out += b"\x00" + _encode_string(name)
        out += _encode_string(value)
    return bytes(out)


# ---- decoder ----------------------------------------------------------------

class Decoder:
    """Connection-scoped HPACK decoder with a dynamic table."""

    def __init__(self, max_table_size: int = 4096):
        self._dynamic: List[Tuple[bytes, bytes]] = []  # newest first
        self._size = 0
        self._max_size = max_table_size
        self._protocol_max = max_table_size

    def _evict(self) -> None:
        while self._size > self._max_size and self._dynamic:
            n, v = self._dynamic.pop()
            self._size -= len(n) + len(v) + 32

    def _add(self, name: bytes, value: bytes) -> None:
        self._dynamic.insert(0, (name, value))
        self._size += len(name) + len(value) + 32
        self._evict()

    def _lookup(self, index: int) -> Tuple[bytes, bytes]:
        if index <= 0:
            raise HpackError("index 0")
        if index <= len(STATIC_TABLE):
            return STATIC_TABLE[index - 1]
        d = index - len(STATIC_TABLE) - 1
        if d >= len(self._dynamic):
            raise HpackError(f"dynamic index {index} out of range")
        return self._dynamic[d]

    def _read_string(self, data: bytes, pos: int) -> Tuple[bytes, int]:
        huff = bool(data[pos] & 0x80)
        length, pos = decode_int(data, pos, 7)
        raw = data[pos : pos + length]
        if len(raw) != length:
            raise HpackError("truncated string")
        pos += length
        return (huffman_decode(raw) if huff else raw), pos

    def decode(self, data: bytes) -> List[Tuple[bytes, bytes]]:
        headers: List[Tuple[bytes, bytes]] = []
        pos = 0
        end = len(data)
        while pos < end:
            b = data[pos]
            if b & 0x80:  # indexed
                index, pos = decode_int(data, pos, 7)
                headers.append(self._lookup(index))
            elif b & 0x40:  # literal with incremental indexing
                index, pos = decode_int(data, pos, 6)
                if index:
                    name = self._lookup(index)[0]
                else:
                    name, pos = self._read_string(data, pos)
                value, pos = self._read_string(data, pos)
                self._add(name, value)
                headers.append((name, value))
            elif b & 0x20:  # dynamic table size update
                size, pos = decode_int(data, pos, 5)
                if size > self._protocol_max:
                    raise HpackError("table size update above SETTINGS limit")
                self._max_size = size
                self._evict()
            else:  # literal without indexing / never indexed (0x00 / 0x10)
                index, pos = decode_int(data, pos, 4)
                if index:
                    name = self._lookup(index)[0]
                else:
                    name, pos = self._read_string(data, pos)
                value, pos = self._read_string(data, pos)
                headers.append((name, value))
        return headers
