"""Minimal self-contained HDF5 writer/reader (no libhdf5, no h5py).

The runtime image ships no h5py, but the reference's output format IS
HDF5 (reference pystella/output.py:52-181 appends to resizable h5py
datasets, and BASELINE.json's north star names "pystella's …
checkpoint layout").  This module writes a spec-compliant subset of
the HDF5 file format directly:

* superblock version 0 (no checksums anywhere in this subset),
* version-1 object headers,
* "old-style" groups (symbol table message -> v1 B-tree + local heap
  + SNOD symbol nodes),
* contiguous datasets of fixed-point / IEEE-float / fixed-ASCII data,
* attributes (scalar and 1-D) on any object.

Files produced here open with stock h5py/libhdf5 (the layout is what
HDF5 1.6-era libraries write).  A matching reader for the same subset
enables round-trip tests in this image.

Datasets are buffered in memory and the file is rewritten on every
flush — append semantics at the API level (like h5py resizable
datasets) with contiguous storage on disk.  Time-series outputs here
are KB-MB scale, so rewrite cost is irrelevant; field checkpoints
write once.
"""

from __future__ import annotations

import struct

import numpy as np

__all__ = ["File", "read_file"]

UNDEF = 0xFFFFFFFFFFFFFFFF


def _pad8(b):
    return b + b"\x00" * (-len(b) % 8)


# ---------------------------------------------------------------------------
# datatype messages

def _datatype_message(dtype):
    """Serialized Datatype message body for a numpy dtype."""
    dt = np.dtype(dtype)
    if dt.kind == "f":
        if dt.itemsize == 8:
            expo_loc, expo_sz, man_sz, bias, prec = 52, 11, 52, 1023, 64
        elif dt.itemsize == 4:
            expo_loc, expo_sz, man_sz, bias, prec = 23, 8, 23, 127, 32
        else:
            raise TypeError(f"unsupported float size {dt.itemsize}")
        # class 1 (float), version 1; LE, mantissa normalization 2
        # (implied MSB), sign bit at position prec-1
        head = struct.pack("<B3BI", (1 << 4) | 1,
                           0x20, prec - 1, 0, dt.itemsize)
        props = struct.pack("<HHBBBBI", 0, prec, expo_loc, expo_sz,
                            0, man_sz, bias)
        return head + props
    if dt.kind in "iu":
        bitfield0 = 0x08 if dt.kind == "i" else 0x00     # signed flag
        head = struct.pack("<B3BI", (1 << 4) | 0,
                           bitfield0, 0, 0, dt.itemsize)
        props = struct.pack("<HH", 0, 8 * dt.itemsize)
        return head + props
    if dt.kind == "S":
        # class 3 (string), null-padded, ASCII
        return struct.pack("<B3BI", (1 << 4) | 3, 0, 0, 0, dt.itemsize)
    raise TypeError(f"unsupported dtype {dt}")


def _parse_datatype(body):
    """Inverse of _datatype_message (subset)."""
    cls = body[0] & 0x0F
    size = struct.unpack("<I", body[4:8])[0]
    if cls == 1:
        return np.dtype(f"<f{size}")
    if cls == 0:
        signed = body[1] & 0x08
        return np.dtype(f"<{'i' if signed else 'u'}{size}")
    if cls == 3:
        return np.dtype(f"S{size}")
    raise TypeError(f"unsupported datatype class {cls}")


def _dataspace_message(shape):
    """Version-1 Dataspace message body (simple or scalar)."""
    rank = len(shape)
    head = struct.pack("<BBB5x", 1, rank, 0)
    dims = b"".join(struct.pack("<Q", n) for n in shape)
    return head + dims


def _parse_dataspace(body):
    rank = body[1]
    flags = body[2]
    dims = struct.unpack(f"<{rank}Q", body[8:8 + 8 * rank])
    if flags & 1:
        pass  # max dims present after dims; ignored
    return dims


# ---------------------------------------------------------------------------
# messages / object headers

def _message(mtype, body):
    body = _pad8(body)
    return struct.pack("<HHB3x", mtype, len(body), 0) + body


def _attr_value(value):
    """Normalize an attribute value -> (numpy array, dtype, shape)."""
    if isinstance(value, str):
        raw = value.encode() or b"\x00"
        dt = np.dtype(f"S{len(raw)}")
        return np.array(raw, dtype=dt), dt, ()
    if isinstance(value, bytes):
        raw = value or b"\x00"
        dt = np.dtype(f"S{len(raw)}")
        return np.array(raw, dtype=dt), dt, ()
    arr = np.asarray(value)
    if arr.dtype.kind == "U":
        return _attr_value(str(value))
    if arr.dtype.kind == "b":
        arr = arr.astype(np.int8)
    if arr.dtype == np.int32:
        arr = arr.astype(np.int64)
    return arr, arr.dtype, arr.shape


def _attribute_message(name, value):
    arr, dt, shape = _attr_value(value)
    nameb = name.encode() + b"\x00"
    dtmsg = _datatype_message(dt)
    dsmsg = _dataspace_message(shape)
    body = struct.pack("<BxHHH", 1, len(nameb), len(dtmsg), len(dsmsg))
    body += _pad8(nameb) + _pad8(dtmsg) + _pad8(dsmsg)
    body += arr.astype(arr.dtype.newbyteorder("<"), copy=False).tobytes()
    return _message(0x000C, body)


def _object_header(messages):
    data = b"".join(messages)
    head = struct.pack("<BxHII4x", 1, len(messages), 1, len(data))
    return head + data


# ---------------------------------------------------------------------------
# writer

class _Piece:
    """A file region whose final address is assigned at assembly time.
    ``fixups`` = [(offset_into_data, piece_or_int_address)]."""

    def __init__(self, data=b""):
        self.data = bytearray(data)
        self.fixups = []
        self.addr = None

    def fix(self, offset, target):
        self.fixups.append((offset, target))


class _Group:
    def __init__(self):
        self.groups = {}       # name -> _Group
        self.datasets = {}     # name -> (ndarray, attrs dict)
        self.attrs = {}

    def require_group(self, name):
        parts = [p for p in name.split("/") if p]
        g = self
        for p in parts:
            if p in g.datasets:
                raise ValueError(f"{p} already a dataset")
            g = g.groups.setdefault(p, _Group())
        return g


class File:
    """Write-mode HDF5 file over the minimal subset.

    API (h5py-flavored)::

        f = File("out.h5")
        f.attrs["argv"] = "..."
        g = f.require_group("energy")
        f.create_dataset("energy/total", data=np.zeros(5))
        f.append("energy/kinetic", row)      # grows axis 0
        f.flush()                            # (re)writes the file
        f.close()
    """

    def __init__(self, filename, mode="w"):
        self.filename = filename
        self.root = _Group()
        self.attrs = self.root.attrs
        self._appends = {}
        self._dirty = True

    # -- h5py-ish surface ------------------------------------------------
    def require_group(self, name):
        self._dirty = True
        return self.root.require_group(name)

    def _resolve(self, path):
        parts = [p for p in path.split("/") if p]
        g = self.root.require_group("/".join(parts[:-1]))
        return g, parts[-1]

    def create_dataset(self, path, data, attrs=None):
        g, name = self._resolve(path)
        arr = np.ascontiguousarray(data)
        if arr.dtype == np.int32:
            arr = arr.astype(np.int64)
        g.datasets[name] = (arr, dict(attrs or {}))
        self._dirty = True

    def append(self, path, row):
        """Append one row along a new leading axis (h5py-resizable
        analogue; reference output.py:157-181)."""
        row = np.asarray(row)
        buf = self._appends.setdefault(path, [])
        buf.append(row)
        self.create_dataset(path, np.stack(buf))

    def __setitem__(self, path, data):
        self.create_dataset(path, data)

    def __getitem__(self, path):
        parts = [p for p in path.split("/") if p]
        g = self.root
        for p in parts[:-1]:
            g = g.groups[p]
        if parts[-1] in g.datasets:
            return g.datasets[parts[-1]][0]
        return g.groups[parts[-1]]

    def flush(self):
        if not self._dirty:
            return
        blob = _assemble(self.root)
        with open(self.filename, "wb") as f:
            f.write(blob)
        self._dirty = False

    def close(self):
        self.flush()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()


def _assemble(root):
    pieces = []

    def add(piece):
        pieces.append(piece)
        return piece

    def build_dataset(arr, attrs):
        raw = add(_Piece(np.ascontiguousarray(arr).astype(
            arr.dtype.newbyteorder("<"), copy=False).tobytes()))
        msgs = [
            _message(0x0001, _dataspace_message(arr.shape)),
            _message(0x0003, _datatype_message(arr.dtype)),
        ]
        layout = struct.pack("<BB", 3, 1) + struct.pack("<QQ", 0, 0)
        lmsg = _message(0x0008, layout)
        for k, v in attrs.items():
            msgs.append(_attribute_message(k, v))
        # find the layout message's address fields after assembly: we
        # place it LAST so its offset inside the header is computable
        msgs.append(lmsg)
        hdr = add(_Piece(_object_header(msgs)))
        # layout body: starts at (header head 16) + sum(len of msgs
        # before) + msg head 8 + 2 bytes (version, class)
        off = 16 + sum(len(m) for m in msgs[:-1]) + 8 + 2
        hdr.fix(off, raw)
        size_off = off + 8
        hdr.data[size_off:size_off + 8] = struct.pack(
            "<Q", arr.nbytes)
        return hdr

    def build_group(g):
        # build children first
        child_hdrs = {}
        for name, sub in g.groups.items():
            child_hdrs[name] = build_group(sub)
        for name, (arr, dattrs) in g.datasets.items():
            child_hdrs[name] = build_dataset(arr, dattrs)

        names = sorted(child_hdrs)
        # local heap: 8 reserved zero bytes, then names
        heap_data = bytearray(b"\x00" * 8)
        name_off = {}
        for n in names:
            name_off[n] = len(heap_data)
            heap_data += n.encode() + b"\x00"
            heap_data += b"\x00" * (-len(heap_data) % 8)
        heap_seg = add(_Piece(bytes(heap_data)))
        heap_hdr = add(_Piece(
            b"HEAP" + struct.pack("<B3x", 0)
            + struct.pack("<QQQ", len(heap_data), 1, 0)))
        heap_hdr.fix(24, heap_seg)      # data segment address

        # SNOD with all entries (sorted); entries_used fits easily for
        # our group sizes (reference outputs have < 32 children)
        snod = bytearray(b"SNOD" + struct.pack("<BxH", 1, len(names)))
        entry_fix = []
        for n in names:
            entry_fix.append((len(snod) + 8, child_hdrs[n]))
            snod += struct.pack("<QQII16x", name_off[n], 0, 0, 0)
        snod_p = add(_Piece(bytes(snod)))
        for off, tgt in entry_fix:
            snod_p.fix(off, tgt)

        # B-tree: one leaf node, 1 child (the SNOD)
        bt = bytearray(b"TREE" + struct.pack("<BBH", 0, 0, 1))
        bt += struct.pack("<QQ", UNDEF, UNDEF)
        bt += struct.pack("<Q", 0)                       # key 0
        child_off = len(bt)
        bt += struct.pack("<Q", 0)                       # child 0 -> SNOD
        bt += struct.pack("<Q", name_off[names[-1]] if names else 0)
        btree = add(_Piece(bytes(bt)))
        btree.fix(child_off, snod_p)

        stab = struct.pack("<QQ", 0, 0)
        msgs = [_message(0x0011, stab)]
        for k, v in g.attrs.items():
            msgs.append(_attribute_message(k, v))
        hdr = add(_Piece(_object_header(msgs)))
        hdr.fix(16 + 8, btree)         # symbol table msg: btree addr
        hdr.fix(16 + 8 + 8, heap_hdr)  # heap addr
        return hdr

    root_hdr = build_group(root)

    # superblock v0
    sb = bytearray(b"\x89HDF\r\n\x1a\n")
    sb += struct.pack("<8B", 0, 0, 0, 0, 0, 8, 8, 0)
    sb += struct.pack("<HHI", 4, 16, 0)
    sb += struct.pack("<QQQQ", 0, UNDEF, 0, UNDEF)   # eof patched below
    # root symbol table entry: name offset 0, header addr, cache 0
    root_entry_off = len(sb) + 8
    sb += struct.pack("<QQII16x", 0, 0, 0, 0)
    sb_piece = _Piece(bytes(sb))
    sb_piece.fix(root_entry_off, root_hdr)
    pieces.insert(0, sb_piece)

    # assign addresses
    addr = 0
    for p in pieces:
        p.addr = addr
        addr += len(p.data) + (-len(p.data) % 8)
    eof = addr

    sb_piece.data[40:48] = struct.pack("<Q", eof)
    out = bytearray()
    for p in pieces:
        for off, tgt in p.fixups:
            a = tgt.addr if isinstance(tgt, _Piece) else int(tgt)
            assert off + 8 <= len(p.data), "fixup outside piece"
            p.data[off:off + 8] = struct.pack("<Q", a)
        out += p.data + b"\x00" * (-len(p.data) % 8)
    assert len(out) == eof
    return bytes(out)


# ---------------------------------------------------------------------------
# reader (same subset; for round-trip tests in this h5py-less image)

def _read_messages(buf, addr):
    ver, nmsg, _refs, hsize = struct.unpack_from("<BxHII", buf, addr)
    assert ver == 1, f"object header v{ver}"
    out = []
    pos = addr + 16
    end = pos + hsize
    while pos < end and len(out) < nmsg:
        mtype, msize, _flags = struct.unpack_from("<HHB3x", buf, pos)
        body = bytes(buf[pos + 8:pos + 8 + msize])
        out.append((mtype, body))
        pos += 8 + msize
        if mtype == 0x0010:           # continuation
            caddr, csize = struct.unpack("<QQ", body[:16])
            pos, end = caddr, caddr + csize
    return out


def _read_attr(body):
    _ver, name_size, dt_size, ds_size = struct.unpack_from("<BxHHH", body)
    off = 8
    name = body[off:off + name_size].split(b"\x00")[0].decode()
    off += name_size + (-name_size % 8)
    dt = _parse_datatype(body[off:off + dt_size])
    off += dt_size + (-dt_size % 8)
    shape = _parse_dataspace(body[off:off + ds_size])
    off += ds_size + (-ds_size % 8)
    count = int(np.prod(shape)) if shape else 1
    val = np.frombuffer(body, dtype=dt, count=count, offset=off)
    if shape:
        val = val.reshape(shape)
    else:
        val = val[0]
        if dt.kind == "S":
            val = val.split(b"\x00")[0].decode()
    return name, val


def _read_object(buf, addr):
    msgs = _read_messages(buf, addr)
    attrs = {}
    shape = dtype = layout = stab = None
    for mtype, body in msgs:
        if mtype == 0x0001:
            shape = _parse_dataspace(body)
        elif mtype == 0x0003:
            dtype = _parse_datatype(body)
        elif mtype == 0x0008:
            assert body[0] == 3 and body[1] == 1, "contiguous v3 only"
            layout = struct.unpack("<QQ", body[2:18])
        elif mtype == 0x0011:
            stab = struct.unpack("<QQ", body[:16])
        elif mtype == 0x000C:
            k, v = _read_attr(body)
            attrs[k] = v

    if stab is not None:
        btree_addr, heap_addr = stab
        assert buf[heap_addr:heap_addr + 4] == b"HEAP"
        heap_seg = struct.unpack_from("<Q", buf, heap_addr + 24)[0]
        assert buf[btree_addr:btree_addr + 4] == b"TREE"
        nent = struct.unpack_from("<H", buf, btree_addr + 6)[0]
        children = {}
        pos = btree_addr + 8 + 16 + 8    # skip head, siblings, key 0
        for _ in range(nent):
            snod_addr = struct.unpack_from("<Q", buf, pos)[0]
            pos += 16                     # child + next key
            assert buf[snod_addr:snod_addr + 4] == b"SNOD"
            nsym = struct.unpack_from("<H", buf, snod_addr + 6)[0]
            epos = snod_addr + 8
            for _s in range(nsym):
                name_off, hdr_addr = struct.unpack_from("<QQ", buf, epos)
                epos += 40
                name = bytes(buf[heap_seg + name_off:]).split(
                    b"\x00")[0].decode()
                children[name] = _read_object(buf, hdr_addr)
        return {"attrs": attrs, "children": children}

    data = None
    if shape is not None and dtype is not None and layout is not None:
        daddr, _dsize = layout
        count = int(np.prod(shape)) if shape else 1
        data = np.frombuffer(buf, dtype=dtype, count=count,
                             offset=daddr).reshape(shape)
    return {"attrs": attrs, "data": data}


def read_file(filename):
    """Parse a (subset-)HDF5 file into nested dicts:
    ``{"attrs": {...}, "children": {name: {...}}}`` with dataset nodes
    ``{"attrs": ..., "data": ndarray}``."""
    with open(filename, "rb") as f:
        buf = f.read()
    assert buf[:8] == b"\x89HDF\r\n\x1a\n", "not an HDF5 file"
    assert buf[8] == 0, f"superblock v{buf[8]} unsupported"
    # root symbol table entry at offset 24 + 32 = 56... superblock v0
    # fixed part is 24 bytes + 4 addresses (32) = 56; entry's object
    # header address is its second 8-byte field
    root_hdr = struct.unpack_from("<Q", buf, 56 + 8)[0]
    return _read_object(buf, root_hdr)
