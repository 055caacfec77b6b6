"""Minimal pure-Python HDF5 (classic format) writer and reader.

The reference checkpoints to `sirius.h5` through the HDF5 C library
(src/core/hdf5_tree.hpp); this stack has neither libhdf5 nor h5py, so
the sirius.h5 tree is emitted directly in the HDF5 v0 classic binary
format: superblock version 0, version-1 object headers, symbol-table
groups (v1 B-tree + local heap + SNOD nodes) and contiguous datasets of
int32/int64/float64.  Files written here are readable by h5py/libhdf5
(standard format), and `read` parses the same subset back for
round-trip tests and restarts.

Layout notes (HDF5 File Format Specification II-III):
- superblock v0: signature + versions + sizes (offsets/lengths = 8) +
  group leaf/internal k + base/eof addresses + root symbol-table entry
- group: object header with a Symbol Table message -> B-tree v1 node
  ("TREE", node type 0) whose children are SNOD symbol nodes; link
  names live in a local heap ("HEAP")
- dataset: object header with Dataspace (simple, v1), Datatype (v1
  fixed-point or IEEE float), and Data Layout (v3, contiguous)
"""

from __future__ import annotations

import struct

import numpy as np

UNDEF = 0xFFFFFFFFFFFFFFFF
LEAF_K = 16          # SNOD holds up to 2*LEAF_K entries


def _pad8(n: int) -> int:
    return (n + 7) & ~7


class _Buf:
    def __init__(self):
        self.data = bytearray()

    def tell(self):
        return len(self.data)

    def write(self, b: bytes):
        self.data += b

    def at(self, pos: int, b: bytes):
        self.data[pos:pos + len(b)] = b

    def align(self, a: int = 8):
        while len(self.data) % a:
            self.data += b"\x00"


def _dtype_message(dt: np.dtype) -> bytes:
    dt = np.dtype(dt)
    if dt == np.dtype("<i4") or dt == np.dtype("<i8"):
        size = dt.itemsize
        # class 0 fixed point, v1; bit3 of bitfield = signed
        head = struct.pack("<B3BI", 0x10, 0x08, 0x00, 0x00, size)
        prop = struct.pack("<HH", 0, 8 * size)
        return head + prop
    if dt == np.dtype("<f8"):
        head = struct.pack("<B3BI", 0x11, 0x20, 0x3F, 0x00, 8)
        prop = struct.pack("<HHBBBBI", 0, 64, 52, 11, 0, 52, 1023)
        return head + prop
    raise TypeError(f"unsupported dtype for HDF5 writer: {dt}")


def _message(mtype: int, body: bytes) -> bytes:
    body_p = body + b"\x00" * (_pad8(len(body)) - len(body))
    return struct.pack("<HHB3x", mtype, len(body_p), 0) + body_p


class H5Writer:
    """Build an HDF5 file as nested dicts then `save(path)`.

    Use `grp(path)["name"] = array` style via `write(path, name, arr)`;
    intermediate groups are created automatically.
    """

    def __init__(self):
        self.tree = {}          # nested dict: str -> dict | np.ndarray

    def create_group(self, path: str):
        node = self.tree
        for part in path.strip("/").split("/"):
            if not part:
                continue
            node = node.setdefault(part, {})
            if not isinstance(node, dict):
                raise ValueError(f"{path}: not a group")
        return node

    def write(self, path: str, name: str, value):
        node = self.create_group(path)
        arr = np.asarray(value)
        if arr.dtype.kind == "i":
            arr = arr.astype("<i4") if arr.itemsize <= 4 else arr.astype("<i8")
        elif arr.dtype.kind == "f":
            arr = arr.astype("<f8")
        elif arr.dtype.kind == "c":
            # store complex as interleaved doubles (reference convention:
            # f_pw written as T* with 2N entries, periodic_function.hpp:183)
            arr = arr.astype("<c16").view("<f8")
        else:
            raise TypeError(f"unsupported array kind {arr.dtype}")
        if arr.ndim == 0:
            arr = arr.reshape(1)
        node[name] = arr

    # ------------------------------------------------------------ emission
    def save(self, path: str):
        buf = _Buf()
        # superblock v0 (96 bytes with 8-byte offsets incl. root entry)
        sb_size = 24 + 8 * 4 + 40
        buf.write(b"\x00" * sb_size)

        root_hdr, root_btree, root_heap = self._emit_group(buf, self.tree)

        eof = buf.tell()
        sb = struct.pack(
            "<8s4B2B2xHHI", b"\x89HDF\r\n\x1a\n",
            0, 0, 0, 0,           # superblock, freespace, root stab, reserved
            0, 8,                 # shared header version, size of offsets
            # oops: order is size_offsets(1), size_lengths(1) after shv
            0, 0, 0)
        # build superblock explicitly (field by field) to avoid confusion
        sb = b"\x89HDF\r\n\x1a\n"
        sb += bytes([0])          # superblock version
        sb += bytes([0])          # free space version
        sb += bytes([0])          # root group symbol table version
        sb += bytes([0])          # reserved
        sb += bytes([0])          # shared header message version
        sb += bytes([8])          # size of offsets
        sb += bytes([8])          # size of lengths
        sb += bytes([0])          # reserved
        sb += struct.pack("<HH", LEAF_K, 16)   # group leaf k, internal k
        sb += struct.pack("<I", 0)             # file consistency flags
        sb += struct.pack("<Q", 0)             # base address
        sb += struct.pack("<Q", UNDEF)         # free space info
        sb += struct.pack("<Q", eof)           # end of file address
        sb += struct.pack("<Q", UNDEF)         # driver info block
        # root group symbol table entry
        sb += struct.pack("<QQI4xQQ", 0, root_hdr, 1, root_btree, root_heap)
        assert len(sb) == sb_size, len(sb)
        buf.at(0, sb)
        with open(path, "wb") as f:
            f.write(bytes(buf.data))

    def _emit_dataset(self, buf: _Buf, arr: np.ndarray) -> int:
        """Write data + object header; return header address."""
        buf.align(8)
        data_addr = buf.tell()
        raw = arr.tobytes()
        buf.write(raw)
        # messages
        dims = arr.shape
        ds_body = struct.pack("<BBB5x", 1, len(dims), 0) \
            + b"".join(struct.pack("<Q", d) for d in dims)
        msgs = _message(0x0001, ds_body)
        msgs += _message(0x0003, _dtype_message(arr.dtype))
        msgs += _message(0x0008, struct.pack("<BBQQ", 3, 1, data_addr,
                                             len(raw)))
        return self._emit_header(buf, msgs, 3)

    def _emit_header(self, buf: _Buf, msgs: bytes, nmsg: int) -> int:
        buf.align(8)
        addr = buf.tell()
        hdr = struct.pack("<BBHI", 1, 0, nmsg, 1) \
            + struct.pack("<I", len(msgs)) + b"\x00" * 4 + msgs
        buf.write(hdr)
        return addr

    def _emit_group(self, buf: _Buf, node: dict):
        """Emit children first, then heap, SNOD(s), B-tree and header.
        Returns (header_addr, btree_addr, heap_addr)."""
        names = sorted(node.keys())
        child_addr = {}
        for name in names:
            v = node[name]
            if isinstance(v, dict):
                h, bt, hp = self._emit_group(buf, v)
                child_addr[name] = (h, bt, hp)
            else:
                child_addr[name] = (self._emit_dataset(buf, v), None, None)

        # local heap: data segment with names ('' at offset 0)
        heap_data = bytearray(b"\x00" * 8)       # empty string + pad
        name_off = {}
        for name in names:
            name_off[name] = len(heap_data)
            nb = name.encode() + b"\x00"
            heap_data += nb
            while len(heap_data) % 8:
                heap_data += b"\x00"
        buf.align(8)
        heap_seg_addr = buf.tell() + 32          # header is 32 bytes
        heap_addr = buf.tell()
        buf.write(b"HEAP" + bytes([0, 0, 0, 0])
                  + struct.pack("<QQQ", len(heap_data), 1, heap_seg_addr))
        buf.write(bytes(heap_data))

        # SNOD nodes (up to 2*LEAF_K entries each)
        cap = 2 * LEAF_K
        chunks = [names[i:i + cap] for i in range(0, len(names), cap)] or [[]]
        snod_addrs = []
        for chunk in chunks:
            buf.align(8)
            a = buf.tell()
            body = b"SNOD" + bytes([1, 0]) + struct.pack("<H", len(chunk))
            for name in chunk:
                h, bt, hp = child_addr[name]
                if bt is not None:
                    ent = struct.pack("<QQI4xQQ", name_off[name], h, 1, bt, hp)
                else:
                    ent = struct.pack("<QQI4x16x", name_off[name], h, 0)
                body += ent
            body += b"\x00" * (8 + 40 * cap - (len(body) - 8))
            buf.write(body)
            snod_addrs.append(a)

        # B-tree v1 node (level 0, children = SNODs)
        buf.align(8)
        btree_addr = buf.tell()
        bt = b"TREE" + bytes([0, 0]) + struct.pack("<H", len(snod_addrs))
        bt += struct.pack("<QQ", UNDEF, UNDEF)
        # keys/children: key0, child0, key1, child1, ..., keyN
        bt += struct.pack("<Q", 0)
        for i, a in enumerate(snod_addrs):
            bt += struct.pack("<Q", a)
            last = chunks[i][-1] if chunks[i] else ""
            bt += struct.pack("<Q", name_off.get(last, 0))
        # pad to max keys for internal_k? Not needed: node size derives
        # from k but readers use entries_used; pad generously anyway
        bt += b"\x00" * (24 + 8 * (2 * LEAF_K + 1) + 16 - len(bt))
        buf.write(bt)

        stab = _message(0x0011, struct.pack("<QQ", btree_addr, heap_addr))
        hdr_addr = self._emit_header(buf, stab, 1)
        return hdr_addr, btree_addr, heap_addr


# ------------------------------------------------------------------- reader
def read(path: str) -> dict:
    """Parse the subset written by H5Writer (and libhdf5 files using
    classic symbol-table groups with contiguous layout)."""
    with open(path, "rb") as f:
        data = f.read()
    assert data[:8] == b"\x89HDF\r\n\x1a\n", "not an HDF5 file"
    # superblock v0: root entry at offset 24+32
    root_entry = 24 + 32
    lnk, hdr, cache, bt, hp = struct.unpack_from("<QQI4xQQ", data, root_entry)
    return _read_group(data, bt, hp)


def _read_group(data: bytes, btree: int, heap: int) -> dict:
    sig = data[heap:heap + 4]
    assert sig == b"HEAP", sig
    seg_size, _free, seg_addr = struct.unpack_from("<QQQ", data, heap + 8)

    def name_at(off):
        end = data.index(b"\x00", seg_addr + off)
        return data[seg_addr + off:end].decode()

    out = {}
    sig = data[btree:btree + 4]
    assert sig == b"TREE", sig
    ntype, level, used = struct.unpack_from("<BBH", data, btree + 4)
    pos = btree + 8 + 16 + 8          # skip siblings + key0
    for i in range(used):
        child = struct.unpack_from("<Q", data, pos)[0]
        pos += 16                      # child + key
        if level > 0:
            out.update(_read_group_btree(data, child, seg_addr))
            continue
        assert data[child:child + 4] == b"SNOD"
        nsym = struct.unpack_from("<H", data, child + 6)[0]
        p = child + 8
        for j in range(nsym):
            lnk, hdr, cache = struct.unpack_from("<QQI", data, p)
            name = name_at(lnk)
            if cache == 1:
                bt2, hp2 = struct.unpack_from("<QQ", data, p + 24)
                out[name] = _read_group(data, bt2, hp2)
            else:
                out[name] = _read_object(data, hdr)
            p += 40
    return out


def _read_object(data: bytes, hdr: int):
    ver, _, nmsg, refc = struct.unpack_from("<BBHI", data, hdr)
    hsize = struct.unpack_from("<I", data, hdr + 8)[0]
    pos = hdr + 16
    end = pos + hsize
    dims = None
    dtype = None
    layout = None
    stab = None
    n = 0
    while pos < end and n < nmsg:
        mtype, msize, flags = struct.unpack_from("<HHB", data, pos)
        body = pos + 8
        if mtype == 0x0001:
            v, ndim, fl = struct.unpack_from("<BBB", data, body)
            off = body + 8 if v == 1 else body + 4
            dims = [struct.unpack_from("<Q", data, off + 8 * i)[0]
                    for i in range(ndim)]
        elif mtype == 0x0003:
            cv = data[body]
            cls = cv & 0x0F
            size = struct.unpack_from("<I", data, body + 4)[0]
            sign = (data[body + 1] >> 3) & 1
            if cls == 0:
                dtype = np.dtype(f"<i{size}" if sign else f"<u{size}")
            elif cls == 1:
                dtype = np.dtype(f"<f{size}")
        elif mtype == 0x0008:
            v = data[body]
            if v == 3:
                lclass = data[body + 1]
                assert lclass == 1, "only contiguous layout supported"
                addr, size = struct.unpack_from("<QQ", data, body + 2)
                layout = (addr, size)
        elif mtype == 0x0011:
            stab = struct.unpack_from("<QQ", data, body)
        pos = body + msize
        n += 1
    if stab is not None:
        return _read_group(data, stab[0], stab[1])
    assert dims is not None and dtype is not None and layout is not None
    addr, size = layout
    arr = np.frombuffer(data[addr:addr + size], dtype=dtype)
    return arr.reshape(dims)


def _read_group_btree(data, child, seg_addr):
    raise NotImplementedError("deep B-trees not produced by this writer")
