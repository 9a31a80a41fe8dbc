"""Read-only parser for BoltDB files (the reference agent's state format).

The reference persists allocations in a BoltDB file with a single bucket
``root`` (ref: pkg/storage/storage.go:13,37-40). BoltDB's on-disk layout is a
public, stable format (magic 0xED0CDAED, version 2): fixed-size pages, two
meta pages, B+tree of branch/leaf pages, small buckets inlined into their
parent leaf element. This module implements just enough of it to enumerate a
bucket's key/value pairs so ``storage.migrate_from_bolt`` can import a node's
existing allocation state without the Go toolchain.
"""
from __future__ import annotations

import struct
from typing import Iterator, List, Tuple

BOLT_MAGIC = 0xED0CDAED
PAGE_HEADER = 16  # id u64, flags u16, count u16, overflow u32
LEAF_ELEM = 16  # flags u32, pos u32, ksize u32, vsize u32
BRANCH_ELEM = 16  # pos u32, ksize u32, pgid u64
FLAG_BRANCH = 0x01
FLAG_LEAF = 0x02
FLAG_META = 0x04
BUCKET_LEAF_FLAG = 0x01
BUCKET_HEADER = 16  # root pgid u64, sequence u64


class BoltFormatError(ValueError):
    pass


def _parse_meta(buf: bytes, off: int) -> dict:
    # meta sits after the page header: magic u32, version u32, page_size u32,
    # flags u32, root{pgid u64, seq u64}, freelist u64, pgid u64, txid u64, checksum u64
    m = struct.unpack_from("<IIII QQ QQQ Q", buf, off + PAGE_HEADER)
    return {
        "magic": m[0],
        "version": m[1],
        "page_size": m[2],
        "root_pgid": m[4],
        "txid": m[8],
    }


def is_bolt_file(path: str) -> bool:
    try:
        with open(path, "rb") as f:
            head = f.read(PAGE_HEADER + 8)
        if len(head) < PAGE_HEADER + 8:
            return False
        magic = struct.unpack_from("<I", head, PAGE_HEADER)[0]
        return magic == BOLT_MAGIC
    except OSError:
        return False


class _BoltFile:
    def __init__(self, data: bytes):
        self.data = data
        metas = []
        for pg in (0, 1):
            try:
                m = _parse_meta(data, pg * 4096)
            except struct.error:
                continue
            if m["magic"] == BOLT_MAGIC:
                metas.append(m)
        if not metas:
            raise BoltFormatError("not a boltdb file (bad magic)")
        meta = max(metas, key=lambda m: m["txid"])
        self.page_size = meta["page_size"]
        # re-read metas at the real page size if it differs from 4096
        if self.page_size != 4096:
            metas = []
            for pg in (0, 1):
                m = _parse_meta(data, pg * self.page_size)
                if m["magic"] == BOLT_MAGIC:
                    metas.append(m)
            meta = max(metas, key=lambda m: m["txid"])
        self.root_pgid = meta["root_pgid"]

    def _page(self, pgid: int) -> Tuple[int, int, int]:
        """Return (offset, flags, count) of page pgid."""
        off = pgid * self.page_size
        _, flags, count, _ = struct.unpack_from("<QHHI", self.data, off)
        return off, flags, count

    def _walk(self, pgid: int) -> Iterator[Tuple[int, bytes, bytes]]:
        """Yield (flags, key, value) of every leaf element under page pgid."""
        off, flags, count = self._page(pgid)
        body = off + PAGE_HEADER
        if flags & FLAG_LEAF:
            for i in range(count):
                eoff = body + i * LEAF_ELEM
                eflags, pos, ksize, vsize = struct.unpack_from("<IIII", self.data, eoff)
                kstart = eoff + pos
                key = self.data[kstart : kstart + ksize]
                val = self.data[kstart + ksize : kstart + ksize + vsize]
                yield eflags, key, val
        elif flags & FLAG_BRANCH:
            for i in range(count):
                eoff = body + i * BRANCH_ELEM
                _pos, _ksize, child = struct.unpack_from("<IIQ", self.data, eoff)
                yield from self._walk(child)
        else:
            raise BoltFormatError(f"unexpected page flags 0x{flags:x} at pgid {pgid}")

    def _inline_walk(self, raw: bytes) -> Iterator[Tuple[int, bytes, bytes]]:
        """Walk an inline bucket: bucket header + a serialized leaf page."""
        page = raw[BUCKET_HEADER:]
        _, flags, count, _ = struct.unpack_from("<QHHI", page, 0)
        if not flags & FLAG_LEAF:
            raise BoltFormatError("inline bucket root is not a leaf")
        for i in range(count):
            eoff = PAGE_HEADER + i * LEAF_ELEM
            eflags, pos, ksize, vsize = struct.unpack_from("<IIII", page, eoff)
            kstart = eoff + pos
            yield eflags, page[kstart : kstart + ksize], page[kstart + ksize : kstart + ksize + vsize]

    def bucket_items(self, name: bytes) -> List[Tuple[bytes, bytes]]:
        for eflags, key, val in self._walk(self.root_pgid):
            if not eflags & BUCKET_LEAF_FLAG or key != name:
                continue
            (root_pgid,) = struct.unpack_from("<Q", val, 0)
            if root_pgid == 0:  # inline bucket
                return [(k, v) for f, k, v in self._inline_walk(val) if not f & BUCKET_LEAF_FLAG]
            return [(k, v) for f, k, v in self._walk(root_pgid) if not f & BUCKET_LEAF_FLAG]
        raise BoltFormatError(f"bucket {name!r} not found")


def read_bolt_bucket(path: str, bucket: bytes) -> List[Tuple[bytes, bytes]]:
    with open(path, "rb") as f:
        data = f.read()
    return _BoltFile(data).bucket_items(bucket)
