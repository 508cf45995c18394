"""Minimal Java-serialization string extractor.

The reference's Binary2Sequence writes SequenceFile KEYS as
ObjectOutputStream-serialized Scala tuples — `(filename, label)` strings
(Binary2Sequence.scala:54-58; SeqImageDataSource.scala:35-63 reads three
tuple layouts).  Full Java deserialization is not needed to consume
them: in the stream grammar (Java Object Serialization Specification
§6.4) every runtime STRING VALUE appears as a TC_STRING (0x74) or
TC_LONGSTRING (0x7C) record, while class/field names inside class
descriptors are written as raw UTF without the tag.  Walking the stream
for tagged strings and dropping JVM type signatures (``Ljava/lang/...;``)
yields the tuple's string payloads in order.
"""

from __future__ import annotations

import struct
from typing import List, Optional

MAGIC = b"\xac\xed"


def is_java_serialized(data: bytes) -> bool:
    return data[:2] == MAGIC


def extract_strings(data: bytes) -> List[str]:
    """All TC_STRING/TC_LONGSTRING payloads in stream order, minus JVM
    type-signature strings."""
    out: List[str] = []
    i = 0
    n = len(data)
    while i < n - 2:
        tag = data[i]
        if tag == 0x74:  # TC_STRING: u16 length + modified-UTF bytes
            ln = struct.unpack(">H", data[i + 1:i + 3])[0]
            raw = data[i + 3:i + 3 + ln]
            if len(raw) == ln:
                try:
                    s = raw.decode("utf-8")
                except UnicodeDecodeError:
                    i += 1
                    continue
                if not (s.startswith("L") and s.endswith(";")) \
                        and not s.startswith("["):
                    out.append(s)
                i += 3 + ln
                continue
        elif tag == 0x7C and i + 9 <= n:  # TC_LONGSTRING: u64 length
            ln = struct.unpack(">Q", data[i + 1:i + 9])[0]
            raw = data[i + 9:i + 9 + ln]
            if len(raw) == ln:
                try:
                    out.append(raw.decode("utf-8"))
                except UnicodeDecodeError:
                    pass
                i += 9 + ln
                continue
        i += 1
    return out


def key_id_label(data: bytes) -> Optional[tuple]:
    """Decode a reference seqfile key: (id, label) from the serialized
    tuple; label parsed as float when possible (their label strings are
    numeric class ids)."""
    if not is_java_serialized(data):
        return None
    strings = extract_strings(data)
    if not strings:
        return None
    sid = strings[0]
    label = 0.0
    if len(strings) > 1:
        try:
            label = float(strings[1])
        except ValueError:
            pass
    return sid, label
