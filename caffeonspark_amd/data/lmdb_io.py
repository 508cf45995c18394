"""Read-only LMDB B-tree walker + bulk writer, pure Python.

The reference reads Caffe-standard LMDB databases of `Datum` protos through
lmdbjni (LmdbRDD.scala:97-155, 240-249).  liblmdb is not available in this
image, so this is a clean-room implementation of the public LMDB on-disk
format (symas.com/lmdb): 4096-byte pages, double meta page, B+tree of
branch/leaf pages, overflow pages for large values.  The writer does a
bulk bottom-up build from sorted key/value pairs — enough for dataset
conversion tools and test fixtures; it is not a transactional store.
"""

from __future__ import annotations

import os
import struct
from typing import Iterator, List, Tuple

PAGESIZE = 4096
PAGEHDRSZ = 16

P_BRANCH = 0x01
P_LEAF = 0x02
P_OVERFLOW = 0x04
P_META = 0x08

F_BIGDATA = 0x01

MDB_MAGIC = 0xBEEFC0DE
MDB_VERSION = 1
P_INVALID = 0xFFFFFFFFFFFFFFFF

_META_DB = struct.Struct("<IHHQQQQQ")  # pad, flags, depth, branch, leaf, ovf, entries, root
_META_HEAD = struct.Struct("<IIQQ")    # magic, version, address, mapsize


def _resolve(path: str) -> str:
    if os.path.isdir(path):
        return os.path.join(path, "data.mdb")
    return path


class LmdbReader:
    def __init__(self, path: str):
        import mmap
        self.path = _resolve(path)
        self._f = open(self.path, "rb")
        # mmap, not read(): partitioned key-range readers touch only the
        # pages of their range (reference LmdbRDD gives each executor a
        # disjoint key range — LmdbRDD.scala:41-95)
        self.data = mmap.mmap(self._f.fileno(), 0, access=mmap.ACCESS_READ)
        self._load_meta()

    def close(self):
        self.data.close()
        self._f.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    def _page(self, pgno: int) -> memoryview:
        off = pgno * PAGESIZE
        return memoryview(self.data)[off:]

    def _load_meta(self):
        best_txn = -1
        for pgno in (0, 1):
            pg = self._page(pgno)
            flags = struct.unpack_from("<H", pg, 10)[0]
            if not (flags & P_META):
                continue
            magic, version, _, _ = _META_HEAD.unpack_from(pg, PAGEHDRSZ)
            if magic != MDB_MAGIC:
                raise ValueError(f"not an LMDB file: {self.path}")
            meta_off = PAGEHDRSZ + _META_HEAD.size
            free_db = _META_DB.unpack_from(pg, meta_off)
            main_db = _META_DB.unpack_from(pg, meta_off + _META_DB.size)
            last_pg, txnid = struct.unpack_from(
                "<QQ", pg, meta_off + 2 * _META_DB.size)
            if txnid > best_txn:
                best_txn = txnid
                self.entries = main_db[6]
                self.root = main_db[7]
                self.depth = main_db[2]
        if best_txn < 0:
            raise ValueError("no valid LMDB meta page")

    def _iter_page(self, pgno: int) -> Iterator[Tuple[bytes, bytes]]:
        pg = self._page(pgno)
        flags, lower = struct.unpack_from("<HH", pg, 10)
        nkeys = (lower - PAGEHDRSZ) >> 1
        if flags & P_BRANCH:
            for i in range(nkeys):
                noff = struct.unpack_from("<H", pg, PAGEHDRSZ + 2 * i)[0]
                lo, hi, nflags, ksize = struct.unpack_from("<HHHH", pg, noff)
                child = lo | (hi << 16) | (nflags << 32)
                yield from self._iter_page(child)
        elif flags & P_LEAF:
            for i in range(nkeys):
                noff = struct.unpack_from("<H", pg, PAGEHDRSZ + 2 * i)[0]
                lo, hi, nflags, ksize = struct.unpack_from("<HHHH", pg, noff)
                dsize = lo | (hi << 16)
                kstart = noff + 8
                key = bytes(pg[kstart:kstart + ksize])
                if nflags & F_BIGDATA:
                    ovpg = struct.unpack_from("<Q", pg, kstart + ksize)[0]
                    opg = self._page(ovpg)
                    yield key, bytes(opg[PAGEHDRSZ:PAGEHDRSZ + dsize])
                else:
                    dstart = kstart + ksize
                    yield key, bytes(pg[dstart:dstart + dsize])

    def items(self) -> Iterator[Tuple[bytes, bytes]]:
        if self.root == P_INVALID or self.entries == 0:
            return
        yield from self._iter_page(self.root)

    # ---- key-range partitioned access (reference LmdbRDD.scala:41-95:
    # partitions are key ranges; each executor reads a disjoint range) ----

    def _branch_children(self, pgno: int) -> List[Tuple[bytes, int]]:
        """(separator_key, child_pgno) pairs of a branch page.  Child i
        holds keys in [key_i, key_{i+1}); key_0 is b'' (= lower bound of
        the parent's range)."""
        pg = self._page(pgno)
        flags, lower = struct.unpack_from("<HH", pg, 10)
        assert flags & P_BRANCH
        nkeys = (lower - PAGEHDRSZ) >> 1
        out = []
        for i in range(nkeys):
            noff = struct.unpack_from("<H", pg, PAGEHDRSZ + 2 * i)[0]
            lo, hi, nflags, ksize = struct.unpack_from("<HHHH", pg, noff)
            child = lo | (hi << 16) | (nflags << 32)
            key = bytes(pg[noff + 8:noff + 8 + ksize])
            out.append((key, child))
        return out

    def _page_flags(self, pgno: int) -> int:
        return struct.unpack_from("<H", self._page(pgno), 10)[0]

    def _iter_range(self, pgno: int, start, end
                    ) -> Iterator[Tuple[bytes, bytes]]:
        """In-order scan of keys in [start, end) touching only pages that
        can intersect the range (start/end None = unbounded)."""
        flags = self._page_flags(pgno)
        if flags & P_BRANCH:
            ch = self._branch_children(pgno)
            for i, (key, child) in enumerate(ch):
                nxt = ch[i + 1][0] if i + 1 < len(ch) else None
                if end is not None and i > 0 and key >= end:
                    break
                if start is not None and nxt is not None and nxt <= start:
                    continue
                yield from self._iter_range(child, start, end)
        elif flags & P_LEAF:
            for key, val in self._iter_page(pgno):
                if start is not None and key < start:
                    continue
                if end is not None and key >= end:
                    break
                yield key, val

    def items_range(self, start=None, end=None
                    ) -> Iterator[Tuple[bytes, bytes]]:
        if self.root == P_INVALID or self.entries == 0:
            return
        yield from self._iter_range(self.root, start, end)

    def partition_ranges(self, n: int) -> List[Tuple[bytes, bytes]]:
        """n (start, end) key ranges covering the whole DB disjointly
        (start None = -inf, end None = +inf).  Split keys come from branch
        separator keys, so discovery reads only branch pages — no full
        scan (the reference discovers ranges with a stride iterator over
        all keys, LmdbRDD.scala:41-95; the B+tree gives them for free)."""
        n = max(1, int(n))
        if self.root == P_INVALID or self.entries == 0:
            return [(None, None)]
        level: List[Tuple[bytes, int]] = [(b"", self.root)]
        while len(level) < n:
            nxt: List[Tuple[bytes, int]] = []
            grew = False
            for key, pgno in level:
                if self._page_flags(pgno) & P_BRANCH:
                    ch = self._branch_children(pgno)
                    # first child inherits the parent's lower bound
                    nxt.append((key, ch[0][1]))
                    nxt.extend(ch[1:])
                    grew = True
                else:
                    nxt.append((key, pgno))
            level = nxt
            if not grew:
                break
        m = len(level)
        n = min(n, m)
        bounds = [level[i * m // n][0] for i in range(n)]
        ranges: List[Tuple[bytes, bytes]] = []
        for i in range(n):
            start = bounds[i] if i > 0 else None
            end = bounds[i + 1] if i + 1 < n else None
            ranges.append((start, end))
        return ranges

    def __len__(self):
        return self.entries


class LmdbWriter:
    """Bulk bottom-up B+tree builder from sorted (key, value) pairs."""

    def __init__(self, path: str, *, subdir: bool = True):
        if subdir:
            os.makedirs(path, exist_ok=True)
            self.path = os.path.join(path, "data.mdb")
        else:
            self.path = _resolve(path)
        self.pages: List[bytes] = [b"\0" * PAGESIZE, b"\0" * PAGESIZE]
        self.n_branch = 0
        self.n_leaf = 0
        self.n_ovf = 0
        self.n_entries = 0

    def _add_page(self, data: bytes) -> int:
        assert len(data) % PAGESIZE == 0
        pgno = len(self.pages)
        for i in range(0, len(data), PAGESIZE):
            self.pages.append(data[i:i + PAGESIZE])
        return pgno

    @staticmethod
    def _page_bytes(pgno: int, flags: int, nodes: List[bytes]) -> bytes:
        lower = PAGEHDRSZ + 2 * len(nodes)
        offs, body = [], b""
        upper = PAGESIZE
        for node in reversed(nodes):
            upper -= len(node)
            offs.append(upper)
            body = node + body
        offs.reverse()
        hdr = struct.pack("<QHHHH", pgno, 0, flags, lower, upper)
        ptrs = b"".join(struct.pack("<H", o) for o in offs)
        pad = b"\0" * (upper - lower)
        page = hdr + ptrs + pad + body
        assert len(page) == PAGESIZE
        return page

    @staticmethod
    def _node(lo: int, hi: int, flags: int, key: bytes,
              data: bytes = b"") -> bytes:
        n = struct.pack("<HHHH", lo, hi, flags, len(key)) + key + data
        if len(n) % 2:
            n += b"\0"
        return n

    def write(self, items: List[Tuple[bytes, bytes]]) -> None:
        items = sorted(items)
        self.n_entries = len(items)
        # ---- leaves
        leaf_space = PAGESIZE - PAGEHDRSZ
        leaves: List[Tuple[bytes, int]] = []  # (first_key, pgno)
        cur_nodes: List[bytes] = []
        cur_used = 0
        cur_first = None

        def flush_leaf():
            nonlocal cur_nodes, cur_used, cur_first
            if not cur_nodes:
                return
            pgno = len(self.pages)
            self.pages.append(None)  # placeholder
            self.pages[pgno] = self._page_bytes(pgno, P_LEAF, cur_nodes)
            leaves.append((cur_first, pgno))
            self.n_leaf += 1
            cur_nodes, cur_used, cur_first = [], 0, None

        for key, val in items:
            if len(key) > 511:
                raise ValueError("LMDB key too long")
            inline_sz = 8 + len(key) + len(val)
            if inline_sz > leaf_space // 2:
                # overflow value
                # overflow chunk: one 16-byte header then contiguous data
                total = PAGEHDRSZ + len(val)
                npages = (total + PAGESIZE - 1) // PAGESIZE
                ovpgno = len(self.pages)
                hdr = struct.pack("<QHHI", ovpgno, 0, P_OVERFLOW, npages)
                blob = hdr + val
                blob += b"\0" * (npages * PAGESIZE - len(blob))
                self._add_page(blob)
                self.n_ovf += npages
                node = self._node(len(val) & 0xFFFF, len(val) >> 16,
                                  F_BIGDATA, key, struct.pack("<Q", ovpgno))
            else:
                node = self._node(len(val) & 0xFFFF, len(val) >> 16, 0, key,
                                  val)
            sz = len(node) + 2  # + ptr slot
            if cur_used + sz > leaf_space:
                flush_leaf()
            if cur_first is None:
                cur_first = key
            cur_nodes.append(node)
            cur_used += sz
        flush_leaf()

        # ---- branches (bottom-up)
        level = leaves
        depth = 1
        while len(level) > 1:
            next_level = []
            cur_nodes, cur_used, cur_first = [], 0, None
            branch_space = PAGESIZE - PAGEHDRSZ

            def flush_branch():
                nonlocal cur_nodes, cur_used, cur_first
                if not cur_nodes:
                    return
                pgno = len(self.pages)
                self.pages.append(None)
                self.pages[pgno] = self._page_bytes(pgno, P_BRANCH, cur_nodes)
                next_level.append((cur_first, pgno))
                self.n_branch += 1
                cur_nodes, cur_used, cur_first = [], 0, None

            for i, (first_key, child) in enumerate(level):
                key = b"" if not cur_nodes else first_key
                node = self._node(child & 0xFFFF, (child >> 16) & 0xFFFF,
                                  (child >> 32) & 0xFFFF, key)
                sz = len(node) + 2
                if cur_used + sz > branch_space:
                    flush_branch()
                    node = self._node(child & 0xFFFF, (child >> 16) & 0xFFFF,
                                      (child >> 32) & 0xFFFF, b"")
                    sz = len(node) + 2
                if cur_first is None:
                    cur_first = first_key
                cur_nodes.append(node)
                cur_used += sz
            flush_branch()
            level = next_level
            depth += 1

        root = level[0][1] if level else P_INVALID
        if not items:
            depth = 0

        # ---- meta pages
        for mp in (0, 1):
            hdr = struct.pack("<QHHHH", mp, 0, P_META, 0, 0)
            head = _META_HEAD.pack(MDB_MAGIC, MDB_VERSION, 0,
                                   len(self.pages) * PAGESIZE)
            free_db = _META_DB.pack(0, 0, 0, 0, 0, 0, 0, P_INVALID)
            main_db = _META_DB.pack(0, 0, depth, self.n_branch, self.n_leaf,
                                    self.n_ovf, self.n_entries, root)
            tail = struct.pack("<QQ", len(self.pages) - 1, 1 + mp)
            page = hdr + head + free_db + main_db + tail
            page += b"\0" * (PAGESIZE - len(page))
            self.pages[mp] = page

        with open(self.path, "wb") as fh:
            for p in self.pages:
                fh.write(p)
