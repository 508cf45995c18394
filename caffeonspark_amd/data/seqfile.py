"""Hadoop SequenceFile reader/writer (uncompressed, BytesWritable pairs).

The reference stores image datasets as sequence files of
(java-serialized key, bytes) pairs (SeqImageDataSource.scala:35-63).  Our
encoding keeps the container format wire-compatible with Hadoop
(SEQ v6, BytesWritable key/value, sync markers) and stores:
  key   = sample id (utf-8)
  value = caffe Datum binary proto (label, dims, encoded flag, bytes)
"""

from __future__ import annotations

import os
import struct
from typing import Iterator, Tuple

_SYNC_INTERVAL = 2000
_CLASS = b"org.apache.hadoop.io.BytesWritable"


def _write_vint(out: bytearray, n: int) -> None:
    # hadoop WritableUtils.writeVInt
    if -112 <= n <= 127:
        out.append(n & 0xFF)
        return
    length = -112
    if n < 0:
        n ^= -1
        length = -120
    tmp = n
    while tmp:
        tmp >>= 8
        length -= 1
    out.append(length & 0xFF)
    size = -(length + 112) if length >= -120 else -(length + 120)
    for i in range(size - 1, -1, -1):
        out.append((n >> (8 * i)) & 0xFF)


def _read_vint(data: bytes, pos: int) -> Tuple[int, int]:
    first = data[pos]
    pos += 1
    if first > 127:
        first -= 256
    if first >= -112:
        return first, pos
    size = -(first + 112) if first >= -120 else -(first + 120)
    neg = first < -120
    n = 0
    for _ in range(size):
        n = (n << 8) | data[pos]
        pos += 1
    return (n ^ -1 if neg else n), pos


class SequenceFileWriter:
    def __init__(self, path: str):
        self.f = open(path, "wb")
        self.sync = os.urandom(16)
        self._since_sync = 0
        hdr = bytearray(b"SEQ\x06")
        _write_vint(hdr, len(_CLASS))
        hdr += _CLASS
        _write_vint(hdr, len(_CLASS))
        hdr += _CLASS
        hdr += b"\x00\x00"              # not compressed, not block
        hdr += struct.pack(">i", 0)     # empty metadata
        hdr += self.sync
        self.f.write(bytes(hdr))

    def append(self, key: bytes, value: bytes) -> None:
        if self._since_sync >= _SYNC_INTERVAL:
            self.f.write(struct.pack(">i", -1))
            self.f.write(self.sync)
            self._since_sync = 0
        krec = struct.pack(">i", len(key)) + key      # BytesWritable
        vrec = struct.pack(">i", len(value)) + value
        rec = struct.pack(">ii", len(krec) + len(vrec), len(krec)) + krec + vrec
        self.f.write(rec)
        self._since_sync += len(rec)

    def close(self):
        self.f.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


class SequenceFileReader:
    def __init__(self, path: str):
        with open(path, "rb") as fh:
            self.data = fh.read()
        if self.data[:3] != b"SEQ":
            raise ValueError(f"not a SequenceFile: {path}")
        self.version = self.data[3]
        pos = 4
        klen, pos = _read_vint(self.data, pos)
        self.key_class = self.data[pos:pos + klen].decode()
        pos += klen
        vlen, pos = _read_vint(self.data, pos)
        self.value_class = self.data[pos:pos + vlen].decode()
        pos += vlen
        self.compressed = self.data[pos] != 0
        self.block_compressed = self.data[pos + 1] != 0
        pos += 2
        if self.compressed or self.block_compressed:
            raise NotImplementedError("compressed SequenceFiles")
        (nmeta,) = struct.unpack_from(">i", self.data, pos)
        pos += 4
        for _ in range(nmeta):
            for _ in range(2):
                ln, pos = _read_vint(self.data, pos)
                pos += ln
        self.sync = self.data[pos:pos + 16]
        self._start = pos + 16

    def items(self) -> Iterator[Tuple[bytes, bytes]]:
        data, pos, n = self.data, self._start, len(self.data)
        while pos < n:
            (rec_len,) = struct.unpack_from(">i", data, pos)
            pos += 4
            if rec_len == -1:  # sync escape
                pos += 16
                continue
            (key_len,) = struct.unpack_from(">i", data, pos)
            pos += 4
            kv = data[pos:pos + rec_len]
            pos += rec_len
            key = kv[4:key_len]          # strip BytesWritable length prefix
            value = kv[key_len + 4:]
            yield key, value
