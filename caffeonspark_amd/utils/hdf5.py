"""Minimal HDF5 (format v0) writer + reader for snapshot files.

Caffe's HDF5 snapshot format (`.caffemodel.h5` / `.solverstate.h5`,
exercised by the reference's cifar10_quick config — SURVEY.md §5
"Checkpoint / resume") stores:
  model:  /data/<layer_name>/<param_idx>  float datasets
  state:  attrs {iter, learned_net, current_step}, group /history/<i>

This module implements the public HDF5 file format's v0 subset needed for
those layouts: superblock v0, v1 group B-trees + local heaps + symbol
nodes, object headers with dataspace/datatype/contiguous-layout messages,
and scalar/string attributes.  Written files follow the spec
(https://support.hdfgroup.org/HDF5/doc/H5.format.html); the reader parses
the same subset.  No h5py/libhdf5 exists in this image, so compatibility
is by-construction from the spec.
"""

from __future__ import annotations

import struct
from typing import Dict, List, Tuple, Union

import numpy as np

UNDEF = 0xFFFFFFFFFFFFFFFF
SIG = b"\x89HDF\r\n\x1a\n"

Node = Union["H5Group", np.ndarray]


class H5Group(dict):
    """name -> H5Group | np.ndarray; .attrs: name -> scalar/str."""

    def __init__(self, *a, **kw):
        super().__init__(*a, **kw)
        self.attrs: Dict[str, Union[int, float, str]] = {}


# =========================================================== writer

class _Writer:
    def __init__(self):
        self.buf = bytearray()

    def tell(self):
        return len(self.buf)

    def write(self, b: bytes) -> int:
        off = len(self.buf)
        self.buf += b
        return off

    def reserve(self, n: int) -> int:
        return self.write(b"\0" * n)

    def patch(self, off: int, b: bytes):
        self.buf[off:off + len(b)] = b


def _dt_message(dtype: np.dtype) -> bytes:
    # class 1 (float) IEEE LE for f4/f8; class 0 (fixed) for ints
    if dtype == np.float32:
        # version 1, class 1 (float); bit field: LE, mantissa-normalized,
        # sign bit 31; properties: offset/precision, exp loc/size,
        # mantissa loc/size, exponent bias
        return struct.pack("<BBBBIHHBBBBI",
                           0x11, 0x20, 0x1F, 0x00, 4,
                           0, 32, 23, 8, 0, 23, 127)
    if dtype == np.float64:
        return struct.pack("<BBBBIHHBBBBI",
                           0x11, 0x20, 0x3F, 0x00, 8,
                           0, 64, 52, 11, 0, 52, 1023)
    if dtype == np.int64:
        return struct.pack("<BBBBIHH",
                           0x10, 0x08, 0x00, 0x00, 8, 0, 64)
    raise TypeError(f"unsupported dtype {dtype}")


def _msg(mtype: int, body: bytes, flags: int = 0) -> bytes:
    if len(body) % 8:
        body += b"\0" * (8 - len(body) % 8)
    return struct.pack("<HHBBBB", mtype, len(body), flags, 0, 0, 0) + body


def _dataspace_body(shape: Tuple[int, ...]) -> bytes:
    body = struct.pack("<BBBB4x", 1, len(shape), 0, 0)
    for d in shape:
        body += struct.pack("<Q", d)
    return body


def _attr_body(name: str, value) -> bytes:
    nb = name.encode() + b"\0"
    if isinstance(value, str):
        vb = value.encode() + b"\0"
        # string datatype: class 3, size = len
        dt = struct.pack("<BBBBI", 0x13, 0x00, 0x00, 0x00, len(vb))
        ds = _dataspace_body(())
        arr = vb
    elif isinstance(value, float):
        dt = _dt_message(np.dtype(np.float64))
        ds = _dataspace_body(())
        arr = struct.pack("<d", value)
    else:
        dt = _dt_message(np.dtype(np.int64))
        ds = _dataspace_body(())
        arr = struct.pack("<q", int(value))

    def pad8(b):
        return b + b"\0" * ((8 - len(b) % 8) % 8)

    body = struct.pack("<BxHHH", 1, len(nb), len(dt), len(ds))
    body += pad8(nb) + pad8(dt) + pad8(ds) + arr
    return body


def _object_header(messages: List[bytes]) -> bytes:
    total = sum(len(m) for m in messages)
    hdr = struct.pack("<BxHII", 1, len(messages), 1, total)
    # v1 object header: align header block to 8 bytes (it already is: 12?)
    # prefix: version(1) pad(1) nmsgs(2) refcount(4) hdrsize(4) + 4 pad
    hdr += b"\0" * 4
    return hdr + b"".join(messages)


def _write_dataset(w: _Writer, arr: np.ndarray) -> int:
    arr = np.ascontiguousarray(arr)
    data_addr = w.write(arr.tobytes())
    msgs = [
        _msg(0x0001, _dataspace_body(arr.shape)),
        _msg(0x0003, _dt_message(arr.dtype)),
        _msg(0x0008, struct.pack("<BB6x", 3, 1) +
             struct.pack("<QQ", data_addr, arr.nbytes)),
    ]
    return w.write(_object_header(msgs))


def _write_group(w: _Writer, group: H5Group) -> int:
    # children first
    child_addrs: List[Tuple[str, int]] = []
    for name, node in group.items():
        if isinstance(node, H5Group):
            addr = _write_group(w, node)
        else:
            addr = _write_dataset(w, np.asarray(node))
        child_addrs.append((name, addr))
    child_addrs.sort()

    # local heap with names
    heap_data = bytearray(b"\0" * 8)  # offset 0 reserved (empty name)
    name_offs = []
    for name, _ in child_addrs:
        name_offs.append(len(heap_data))
        heap_data += name.encode() + b"\0"
        while len(heap_data) % 8:
            heap_data += b"\0"
    heap_data_addr = w.write(bytes(heap_data))
    heap_addr = w.write(b"HEAP" + struct.pack("<B3xQQQ", 0, len(heap_data),
                                              UNDEF, heap_data_addr))

    # symbol node with entries
    snod = b"SNOD" + struct.pack("<BxH", 1, len(child_addrs))
    for (name, addr), noff in zip(child_addrs, name_offs):
        snod += struct.pack("<QQII", noff, addr, 0, 0) + b"\0" * 16
    snod_addr = w.write(snod)

    # B-tree v1 leaf pointing at the symbol node
    btree = b"TREE" + struct.pack("<BBH", 0, 0, 1 if child_addrs else 0)
    btree += struct.pack("<QQ", UNDEF, UNDEF)
    # keys/children: key0 (heap off 0), child snod, key1 (last name off)
    btree += struct.pack("<Q", 0)
    if child_addrs:
        btree += struct.pack("<Q", snod_addr)
        btree += struct.pack("<Q", name_offs[-1])
    btree_addr = w.write(btree)

    msgs = [_msg(0x0011, struct.pack("<QQ", btree_addr, heap_addr))]
    for aname, aval in group.attrs.items():
        msgs.append(_msg(0x000C, _attr_body(aname, aval)))
    return w.write(_object_header(msgs))


def save(path: str, root: H5Group) -> None:
    w = _Writer()
    w.reserve(96)  # superblock v0 (56) + root symbol table entry (40)
    root_hdr = _write_group(w, root)
    # sig(8) versions/sizes(8) ks(4+4... spec: leaf k, internal k, flags)
    sb = SIG
    sb += struct.pack("<BBBBBBBB", 0, 0, 0, 0, 0, 8, 8, 0)
    sb += struct.pack("<HHI", 4, 16, 0)
    sb += struct.pack("<QQQQ", 0, UNDEF, len(w.buf), UNDEF)
    sb += struct.pack("<QQII", 0, root_hdr, 0, 0) + b"\0" * 16
    assert len(sb) == 96
    w.patch(0, sb)
    with open(path, "wb") as fh:
        fh.write(bytes(w.buf))


# =========================================================== reader

class _Reader:
    def __init__(self, data: bytes):
        self.d = data
        if data[:8] != SIG:
            raise ValueError("not an HDF5 file")
        # superblock v0: sizes at fixed offsets for our subset
        self.root_hdr = struct.unpack_from("<Q", data, 8 + 16 + 32 + 8)[0]

    def read_object(self, addr: int):
        d = self.d
        ver, nmsgs, refcount, hdrsize = struct.unpack_from("<BxHII", d, addr)
        pos = addr + 16
        end = pos + hdrsize
        messages = []
        while pos < end - 7:
            mtype, size, flags = struct.unpack_from("<HHB", d, pos)
            body = d[pos + 8:pos + 8 + size]
            messages.append((mtype, body))
            pos += 8 + size
        return messages

    def parse(self, addr: int):
        msgs = self.read_object(addr)
        types = {t for t, _ in msgs}
        if 0x0011 in types:  # symbol table -> group
            grp = H5Group()
            for t, body in msgs:
                if t == 0x0011:
                    btree_addr, heap_addr = struct.unpack_from("<QQ", body, 0)
                    for name, child in self._iter_group(btree_addr,
                                                        heap_addr):
                        grp[name] = self.parse(child)
                elif t == 0x000C:
                    name, val = self._parse_attr(body)
                    grp.attrs[name] = val
            return grp
        # dataset
        shape, dtype, data_addr, nbytes = None, None, None, 0
        for t, body in msgs:
            if t == 0x0001:
                ver, ndim = struct.unpack_from("<BB", body, 0)
                shape = struct.unpack_from(f"<{ndim}Q", body, 8)
            elif t == 0x0003:
                cv = body[0]
                cls = cv & 0x0F
                size = struct.unpack_from("<I", body, 4)[0]
                if cls == 1:
                    dtype = np.float32 if size == 4 else np.float64
                elif cls == 0:
                    dtype = np.int64
                else:
                    raise TypeError(f"dataset datatype class {cls}")
            elif t == 0x0008:
                ver, lclass = struct.unpack_from("<BB", body, 0)
                data_addr, nbytes = struct.unpack_from("<QQ", body, 8)
        arr = np.frombuffer(self.d, dtype=dtype,
                            count=nbytes // np.dtype(dtype).itemsize,
                            offset=data_addr).reshape(shape)
        return arr.copy()

    def _iter_group(self, btree_addr: int, heap_addr: int):
        d = self.d
        assert d[btree_addr:btree_addr + 4] == b"TREE"
        node_type, level, entries = struct.unpack_from("<BBH", d,
                                                       btree_addr + 4)
        heap_data_addr = struct.unpack_from("<Q", d, heap_addr + 24)[0]
        pos = btree_addr + 8 + 16  # skip siblings
        pos += 8  # key0
        for _ in range(entries):
            child = struct.unpack_from("<Q", d, pos)[0]
            pos += 16  # child + next key
            if level > 0:
                yield from self._iter_group(child, heap_addr)
                continue
            assert d[child:child + 4] == b"SNOD"
            nsyms = struct.unpack_from("<H", d, child + 6)[0]
            epos = child + 8
            for _ in range(nsyms):
                name_off, hdr = struct.unpack_from("<QQ", d, epos)
                name_pos = heap_data_addr + name_off
                name_end = d.index(b"\0", name_pos)
                yield d[name_pos:name_end].decode(), hdr
                epos += 40

    def _parse_attr(self, body: bytes):
        ver, name_size, dt_size, ds_size = struct.unpack_from("<BxHHH", body,
                                                              0)
        def pad8(n):
            return n + ((8 - n % 8) % 8)
        pos = 8
        name = body[pos:pos + name_size].rstrip(b"\0").decode()
        pos += pad8(name_size)
        dt = body[pos:pos + dt_size]
        pos += pad8(dt_size)
        pos += pad8(ds_size)
        cls = dt[0] & 0x0F
        if cls == 3:
            size = struct.unpack_from("<I", dt, 4)[0]
            return name, body[pos:pos + size].rstrip(b"\0").decode()
        if cls == 1:
            return name, struct.unpack_from("<d", body, pos)[0]
        return name, struct.unpack_from("<q", body, pos)[0]


def load(path: str) -> H5Group:
    with open(path, "rb") as fh:
        r = _Reader(fh.read())
    return r.parse(r.root_hdr)


# ================================================== caffe snapshot layer

def save_net(path: str, net) -> None:
    """Model -> /data/<layer>/<param_idx> (caffe NetParameter HDF5 layout)."""
    root = H5Group()
    data = H5Group()
    root["data"] = data
    for layer in net.layers:
        if not layer.blobs:
            continue
        lg = H5Group()
        for i, b in enumerate(layer.blobs):
            lg[str(i)] = b.data.detach().cpu().float().numpy()
        data[layer.name] = lg
    save(path, root)


def load_net(path: str, net) -> None:
    root = load(path)
    data = root.get("data", root)
    for layer in net.layers:
        lg = data.get(layer.name)
        if lg is None:
            continue
        for i, b in enumerate(layer.blobs):
            arr = lg.get(str(i))
            if arr is None:
                continue
            import torch
            t = torch.from_numpy(np.ascontiguousarray(arr))
            b.data.copy_(t.reshape(b.data.shape).to(b.data.dtype))


def save_solver_state(path: str, solver, learned_net: str) -> None:
    root = H5Group()
    root.attrs["iter"] = int(solver.iter)
    root.attrs["current_step"] = int(solver.current_step)
    root.attrs["learned_net"] = learned_net
    hist = H5Group()
    root["history"] = hist
    for i, h in enumerate(solver.history + solver.history2):
        hist[str(i)] = h.detach().cpu().float().numpy()
    save(path, root)


def load_solver_state(path: str, solver) -> None:
    import os

    import torch
    root = load(path)
    solver.iter = int(root.attrs.get("iter", 0))
    solver.current_step = int(root.attrs.get("current_step", 0))
    hist = root.get("history", H5Group())
    for i, h in enumerate(solver.history + solver.history2):
        arr = hist.get(str(i))
        if arr is not None:
            h.copy_(torch.from_numpy(np.ascontiguousarray(arr))
                    .reshape(h.shape).to(h.dtype))
    learned = root.attrs.get("learned_net", "")
    if learned and os.path.exists(learned):
        solver.load_weights(learned)


def set_learned_net(path: str, learned_net: str) -> None:
    """Rewrite the learned_net attribute (reference: CaffeNet::setLearnedNet
    HDF5 path, CaffeNet.cpp:344-356)."""
    root = load(path)
    root.attrs["learned_net"] = learned_net
    save(path, root)
