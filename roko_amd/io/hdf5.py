"""Minimal pure-Python HDF5 reader/writer for the reference's dataset layout.

The reference stores features in HDF5 via h5py 2.10 (reference
roko/data.py:29-48: groups ``{contig}_{start}-{end}`` with datasets
``positions``/``labels``/``examples`` and a ``contigs`` group of per-contig
attrs). This environment has no h5py, so interop is implemented directly
against the HDF5 file format spec:

  * reader: superblock v0, v1 object headers (+ continuations), old-style
    (symbol-table) groups, contiguous AND chunked datasets (v1 chunk
    B-trees, optional gzip/shuffle filters), attributes incl. h5py's
    variable-length strings (global heap) — everything h5py 2.10 emits for
    the reference layout;
  * writer: superblock v0, v1 object headers, symbol-table groups,
    contiguous datasets, fixed/vlen string + integer attributes — a subset
    that h5py (any version) opens.

Scope: what the roko interop needs, not a general HDF5 implementation —
unsupported constructs raise with a clear message.
"""

from __future__ import annotations

import struct
import zlib
from typing import Dict, List, Optional, Tuple, Union

import numpy as np

UNDEF = 0xFFFFFFFFFFFFFFFF
_SIG = b"\x89HDF\r\n\x1a\n"


# ===========================================================================
# Reader
# ===========================================================================

class _Buf:
    def __init__(self, data: bytes):
        self.d = data

    def u8(self, o): return self.d[o]
    def u16(self, o): return struct.unpack_from("<H", self.d, o)[0]
    def u32(self, o): return struct.unpack_from("<I", self.d, o)[0]
    def u64(self, o): return struct.unpack_from("<Q", self.d, o)[0]


class H5Dtype:
    """Parsed datatype message."""

    def __init__(self, cls: int, size: int, bits0: int, np_dtype=None,
                 vlen_string: bool = False):
        self.cls = cls
        self.size = size
        self.bits0 = bits0
        self.np_dtype = np_dtype
        self.vlen_string = vlen_string


def _parse_datatype(b: _Buf, o: int) -> H5Dtype:
    cv = b.u8(o)
    cls = cv & 0x0F
    bits0 = b.u8(o + 1)
    bits1 = b.u8(o + 2)
    size = b.u32(o + 4)
    if cls == 0:  # fixed point
        signed = bool(bits0 & 0x08)
        be = bool(bits0 & 0x01)
        base = {1: "i1", 2: "i2", 4: "i4", 8: "i8"}[size] if signed else \
               {1: "u1", 2: "u2", 4: "u4", 8: "u8"}[size]
        return H5Dtype(cls, size, bits0, np.dtype((">" if be else "<") + base))
    if cls == 1:  # float
        be = bool(bits0 & 0x01)
        base = {4: "f4", 8: "f8"}[size]
        return H5Dtype(cls, size, bits0, np.dtype((">" if be else "<") + base))
    if cls == 3:  # fixed string
        return H5Dtype(cls, size, bits0, np.dtype(f"S{size}"))
    if cls == 9:  # variable length
        vtype = bits0 & 0x0F
        return H5Dtype(cls, size, bits0, None, vlen_string=(vtype == 1))
    raise NotImplementedError(f"HDF5 datatype class {cls} not supported")


def _parse_dataspace(b: _Buf, o: int) -> Tuple[int, ...]:
    ver = b.u8(o)
    if ver == 1:
        nd = b.u8(o + 1)
        # flags at o+2; dims start at o+8
        return tuple(b.u64(o + 8 + 8 * i) for i in range(nd))
    if ver == 2:
        nd = b.u8(o + 1)
        # version 2: version, dims, flags, type, then dims
        return tuple(b.u64(o + 4 + 8 * i) for i in range(nd))
    raise NotImplementedError(f"dataspace version {ver}")


class H5Object:
    """One object header: messages collected, attrs parsed."""

    def __init__(self, f: "H5File", addr: int):
        self.f = f
        self.addr = addr
        self.msgs: List[Tuple[int, int, int]] = []  # (type, offset, size)
        self._parse_v1(addr)
        self.attrs: Dict[str, object] = {}
        for t, o, s in self.msgs:
            if t == 0x000C:
                name, val = self._parse_attr(o)
                self.attrs[name] = val

    def _parse_v1(self, addr: int) -> None:
        b = self.f.b
        ver = b.u8(addr)
        if ver != 1:
            raise NotImplementedError(
                f"object header version {ver} (write the file with h5py "
                "default/earliest libver)")
        nmsg = b.u16(addr + 2)
        blocks = [(addr + 16, b.u32(addr + 8))]
        seen = 0
        while blocks and seen < nmsg:
            o, size = blocks.pop(0)
            end = o + size
            while o + 8 <= end and seen < nmsg:
                t = b.u16(o)
                s = b.u16(o + 2)
                body = o + 8
                if t == 0x0010:  # continuation
                    blocks.append((b.u64(body), b.u64(body + 8)))
                elif t != 0x0000:
                    self.msgs.append((t, body, s))
                seen += 1
                o = body + s

    def _msg(self, mtype: int) -> Optional[Tuple[int, int]]:
        for t, o, s in self.msgs:
            if t == mtype:
                return o, s
        return None

    # -- attributes ---------------------------------------------------------
    def _parse_attr(self, o: int):
        b = self.f.b
        ver = b.u8(o)
        if ver == 1:
            name_size = b.u16(o + 2)
            dt_size = b.u16(o + 4)
            ds_size = b.u16(o + 6)
            p = o + 8
            name = b.d[p:p + name_size].split(b"\x00")[0].decode("utf-8")
            p += (name_size + 7) & ~7
            dt = _parse_datatype(b, p)
            p += (dt_size + 7) & ~7
            shape = _parse_dataspace(b, p)
            p += (ds_size + 7) & ~7
        elif ver in (2, 3):
            name_size = b.u16(o + 2)
            dt_size = b.u16(o + 4)
            ds_size = b.u16(o + 6)
            p = o + 8
            if ver == 3:
                p += 1  # name character-set encoding
            name = b.d[p:p + name_size].split(b"\x00")[0].decode("utf-8")
            p += name_size  # v2/3: not padded
            dt = _parse_datatype(b, p)
            p += dt_size
            shape = _parse_dataspace(b, p)
            p += ds_size
        else:
            raise NotImplementedError(f"attribute message version {ver}")
        n = int(np.prod(shape)) if shape else 1
        val = self._read_typed(p, dt, n)
        if shape == ():
            return name, val[0] if isinstance(val, (list, np.ndarray)) else val
        return name, val

    def _read_typed(self, p: int, dt: H5Dtype, n: int):
        b = self.f.b
        if dt.vlen_string:
            out = []
            for i in range(n):
                q = p + 16 * i
                length = b.u32(q)
                gaddr = b.u64(q + 4)
                gidx = b.u32(q + 12)
                out.append(self.f.gheap_object(gaddr, gidx)[:length]
                           .decode("utf-8"))
            return out[0] if n == 1 else out
        if dt.cls == 3:
            raw = b.d[p:p + dt.size * n]
            if n == 1:
                return raw[:dt.size].split(b"\x00")[0].decode("utf-8",
                                                              "replace")
            return [raw[i * dt.size:(i + 1) * dt.size].split(b"\x00")[0]
                    .decode("utf-8", "replace") for i in range(n)]
        arr = np.frombuffer(b.d, dtype=dt.np_dtype, count=n, offset=p)
        if n == 1:
            return arr[0].item()
        return arr.copy()


class H5Dataset(H5Object):
    def __init__(self, f: "H5File", addr: int, name: str):
        super().__init__(f, addr)
        self.name = name
        m = self._msg(0x0001)
        self.shape = _parse_dataspace(f.b, m[0]) if m else ()
        m = self._msg(0x0003)
        self.dtype_info = _parse_datatype(f.b, m[0])
        self.dtype = self.dtype_info.np_dtype
        self._data: Optional[np.ndarray] = None

    # filters: list of (id,) — gzip=1, shuffle=2
    def _filters(self) -> List[int]:
        m = self._msg(0x000B)
        if m is None:
            return []
        b, (o, _) = self.f.b, m
        ver = b.u8(o)
        nf = b.u8(o + 1)
        ids = []
        p = o + (8 if ver == 1 else 2)
        for _ in range(nf):
            fid = b.u16(p)
            name_len = b.u16(p + 2) if ver == 1 else (0 if fid < 256 else b.u16(p + 2))
            ncd = b.u16(p + 6)
            p += 8 + name_len + 2 * ncd
            if ver == 1 and ncd % 2:
                p += 2
            ids.append(fid)
        return ids

    def _read(self) -> np.ndarray:
        if self._data is not None:
            return self._data
        b = self.f.b
        m = self._msg(0x0008)
        if m is None:
            raise ValueError(f"dataset {self.name}: no layout message")
        o, _ = m
        ver = b.u8(o)
        if ver != 3:
            raise NotImplementedError(f"data layout version {ver}")
        lclass = b.u8(o + 1)
        nbytes = int(np.prod(self.shape)) * self.dtype.itemsize
        if lclass == 1:  # contiguous
            addr = b.u64(o + 2)
            if addr == UNDEF:
                arr = np.zeros(self.shape, self.dtype)
            else:
                arr = np.frombuffer(b.d, self.dtype,
                                    count=int(np.prod(self.shape)),
                                    offset=addr).reshape(self.shape)
        elif lclass == 2:  # chunked
            nd1 = b.u8(o + 2)  # dimensionality + 1
            btree = b.u64(o + 3)
            chunk = tuple(b.u32(o + 11 + 4 * i) for i in range(nd1 - 1))
            arr = np.zeros(self.shape, self.dtype)
            if btree != UNDEF:
                self._read_chunks(btree, arr, chunk, self._filters())
        elif lclass == 0:  # compact
            size = b.u16(o + 2)
            arr = np.frombuffer(b.d, self.dtype,
                                count=int(np.prod(self.shape)),
                                offset=o + 4).reshape(self.shape)
        else:
            raise NotImplementedError(f"layout class {lclass}")
        del nbytes
        self._data = arr
        return arr

    def _read_chunks(self, node: int, arr: np.ndarray,
                     chunk: Tuple[int, ...], filters: List[int]) -> None:
        b = self.f.b
        if b.d[node:node + 4] != b"TREE":
            raise ValueError("bad chunk B-tree node")
        level = b.u8(node + 5)
        nent = b.u16(node + 6)
        nd = len(chunk)
        keysize = 8 + 8 * (nd + 1)
        p = node + 24
        for i in range(nent):
            csize = b.u32(p)
            # filter mask at p+4
            offs = tuple(b.u64(p + 8 + 8 * j) for j in range(nd))
            child = b.u64(p + keysize)
            if level > 0:
                self._read_chunks(child, arr, chunk, filters)
            else:
                raw = b.d[child:child + csize]
                if 1 in filters:
                    raw = zlib.decompress(raw)
                if 2 in filters:  # shuffle: de-interleave bytes
                    it = self.dtype.itemsize
                    n = len(raw) // it
                    raw = (np.frombuffer(raw, np.uint8).reshape(it, n)
                           .T.tobytes())
                cdata = np.frombuffer(raw, self.dtype,
                                      count=int(np.prod(chunk))).reshape(chunk)
                sl = tuple(slice(offs[j], min(offs[j] + chunk[j],
                                              arr.shape[j]))
                           for j in range(nd))
                csl = tuple(slice(0, sl[j].stop - sl[j].start)
                            for j in range(nd))
                arr[sl] = cdata[csl]
            p += keysize + 8

    def __getitem__(self, idx):
        return self._read()[idx]

    def __len__(self):
        return self.shape[0] if self.shape else 0

    def __array__(self, dtype=None):
        a = self._read()
        return a.astype(dtype) if dtype is not None else a


class H5Group(H5Object):
    def __init__(self, f: "H5File", addr: int, name: str = "/"):
        super().__init__(f, addr)
        self.name = name
        self._entries: Dict[str, Tuple[int, bool]] = {}  # name -> (addr, ?)
        m = self._msg(0x0011)
        if m is not None:
            btree = f.b.u64(m[0])
            heap = f.b.u64(m[0] + 8)
            if btree != UNDEF:
                self._walk_btree(btree, heap)

    def _walk_btree(self, node: int, heap: int) -> None:
        b = self.f.b
        if b.d[node:node + 4] != b"TREE":
            raise ValueError("bad group B-tree node")
        level = b.u8(node + 5)
        nent = b.u16(node + 6)
        p = node + 24
        for i in range(nent):
            child = b.u64(p + 8)
            if level > 0:
                self._walk_btree(child, heap)
            else:
                self._walk_snod(child, heap)
            p += 16

    def _walk_snod(self, snod: int, heap: int) -> None:
        b = self.f.b
        if b.d[snod:snod + 4] != b"SNOD":
            raise ValueError("bad symbol node")
        n = b.u16(snod + 6)
        heap_data = b.u64(heap + 24)
        for i in range(n):
            e = snod + 8 + 40 * i
            name_off = b.u64(e)
            addr = b.u64(e + 8)
            name = b.d[heap_data + name_off:
                       heap_data + name_off + 1024].split(b"\x00")[0]
            self._entries[name.decode("utf-8")] = (addr, True)

    def keys(self):
        return list(self._entries)

    def __contains__(self, k):
        return k in self._entries

    def __getitem__(self, k: str) -> Union["H5Group", H5Dataset]:
        addr, _ = self._entries[k]
        obj = H5Object(self.f, addr)
        if obj._msg(0x0011) is not None:
            return H5Group(self.f, addr, k)
        return H5Dataset(self.f, addr, k)


class H5File:
    """Read-only HDF5 file (see module docstring for scope)."""

    def __init__(self, path: str):
        with open(path, "rb") as fh:
            data = fh.read()
        if data[:8] != _SIG:
            raise ValueError(f"{path}: not an HDF5 file")
        if len(data) < 96:
            raise ValueError(f"{path}: truncated HDF5 file "
                             f"({len(data)} bytes, superblock needs 96)")
        self.b = _Buf(data)
        ver = self.b.u8(8)
        if ver not in (0, 1):
            raise NotImplementedError(
                f"superblock version {ver}: re-save with h5py libver="
                "'earliest' for interop")
        if self.b.u8(13) != 8 or self.b.u8(14) != 8:
            raise NotImplementedError("only 8-byte offsets/lengths supported")
        # root symbol table entry at 24 + 32 = 56 (v0) / 56+? (v1 adds 4)
        root_entry = 24 + 32 + (4 if ver == 1 else 0)
        root_addr = self.b.u64(root_entry + 8)
        self.root = H5Group(self, root_addr, "/")
        self._gheaps: Dict[int, Dict[int, bytes]] = {}

    def keys(self):
        return self.root.keys()

    def __contains__(self, k):
        return k in self.root

    def __getitem__(self, k):
        return self.root[k]

    @property
    def attrs(self):
        return self.root.attrs

    def gheap_object(self, addr: int, idx: int) -> bytes:
        if addr not in self._gheaps:
            b = self.b
            if b.d[addr:addr + 4] != b"GCOL":
                raise ValueError("bad global heap collection")
            size = b.u64(addr + 8)
            objs: Dict[int, bytes] = {}
            p = addr + 16
            end = addr + size
            while p + 16 <= end:
                oidx = b.u16(p)
                osize = b.u64(p + 8)
                if oidx == 0:
                    break
                objs[oidx] = b.d[p + 16:p + 16 + osize]
                p += 16 + ((osize + 7) & ~7)
            self._gheaps[addr] = objs
        return self._gheaps[addr][idx]


# ===========================================================================
# Writer
# ===========================================================================

def _pad8(n: int) -> int:
    return (n + 7) & ~7


class _W:
    """Append-only file image with 8-byte-aligned allocation."""

    def __init__(self):
        self.parts: List[bytes] = []
        self.off = 0

    def alloc(self, data: bytes) -> int:
        pad = _pad8(self.off) - self.off
        if pad:
            self.parts.append(b"\x00" * pad)
            self.off += pad
        addr = self.off
        self.parts.append(data)
        self.off += len(data)
        return addr


def _dt_msg(dtype: np.dtype) -> bytes:
    """Datatype message body for integer/float/fixed-string numpy dtypes."""
    dtype = np.dtype(dtype)
    if dtype.kind in "iu":
        bits0 = 0x08 if dtype.kind == "i" else 0x00
        return struct.pack("<BBBBIHH", 0x10, bits0, 0, 0, dtype.itemsize,
                           0, dtype.itemsize * 8)
    if dtype.kind == "f":
        # IEEE little-endian: class 1, standard bit fields + properties
        if dtype.itemsize == 4:
            return struct.pack("<BBBBIHHBBBBI", 0x11, 0x20, 0x3F, 0, 4,
                               0, 32, 23, 8, 0, 23, 127)
        return struct.pack("<BBBBIHHBBBBI", 0x11, 0x20, 0x3F, 0, 8,
                           0, 64, 52, 11, 0, 52, 1023)
    if dtype.kind == "S":
        # fixed ASCII string, null-padded
        return struct.pack("<BBBBI", 0x13, 0x00, 0, 0, dtype.itemsize)
    raise NotImplementedError(f"writer: dtype {dtype}")


_VLEN_STR_DT = (struct.pack("<BBBBI", 0x19, 0x01, 0, 0, 16)
                + struct.pack("<BBBBI", 0x13, 0x00, 0, 0, 1))


def _ds_msg(shape: Tuple[int, ...]) -> bytes:
    body = struct.pack("<BBBBI", 1, len(shape), 0, 0, 0)
    for s in shape:
        body += struct.pack("<Q", s)
    return body


class _Msg:
    def __init__(self, mtype: int, body: bytes):
        self.mtype = mtype
        self.body = body


def _attr_msg(name: str, value) -> _Msg:
    nb = name.encode("utf-8") + b"\x00"
    if isinstance(value, str):
        data = value.encode("utf-8")
        dt = struct.pack("<BBBBI", 0x13, 0x00, 0, 0, max(1, len(data)))
        ds = _ds_msg(())
    elif isinstance(value, (int, np.integer)):
        data = struct.pack("<q", int(value))
        dt = _dt_msg(np.dtype("<i8"))
        ds = _ds_msg(())
    elif isinstance(value, (float, np.floating)):
        data = struct.pack("<d", float(value))
        dt = _dt_msg(np.dtype("<f8"))
        ds = _ds_msg(())
    else:
        raise NotImplementedError(f"attr type {type(value)}")
    body = struct.pack("<BBHHH", 1, 0, len(nb), len(dt), len(ds))
    body += nb + b"\x00" * (_pad8(len(nb)) - len(nb))
    body += dt + b"\x00" * (_pad8(len(dt)) - len(dt))
    body += ds + b"\x00" * (_pad8(len(ds)) - len(ds))
    body += data
    return _Msg(0x000C, body)


def _object_header(msgs: List[_Msg]) -> bytes:
    blob = b""
    for m in msgs:
        body = m.body + b"\x00" * (_pad8(len(m.body)) - len(m.body))
        blob += struct.pack("<HHBBBB", m.mtype, len(body), 0, 0, 0, 0) + body
    return struct.pack("<BBHII", 1, 0, len(msgs), 1, len(blob)) + b"\x00" * 4 + blob


class WGroup:
    def __init__(self, writer: "H5Writer", name: str):
        self.writer = writer
        self.name = name
        self.attrs: Dict[str, object] = {}
        self.children: Dict[str, object] = {}  # name -> WGroup | WDataset

    def create_group(self, name: str) -> "WGroup":
        g = WGroup(self.writer, name)
        self.children[name] = g
        return g

    def create_dataset(self, name: str, data: np.ndarray,
                       chunks: Optional[Tuple[int, ...]] = None,
                       **kw) -> "WDataset":
        d = WDataset(name, np.ascontiguousarray(data), chunks=chunks)
        self.children[name] = d
        return d

    def __setitem__(self, name: str, data) -> None:
        self.create_dataset(name, np.asarray(data))

    # -- serialization ------------------------------------------------------
    # B-tree v1 capacities from the superblock: symbol nodes hold up to
    # 2*group_leaf_k = 8 entries, btree nodes up to 2*group_internal_k = 32
    # children; readers fetch FULL-capacity node images, so nodes are padded.
    _SNOD_CAP = 8
    _BT_CAP = 32
    _SNOD_SIZE = 8 + 8 * 40
    _BT_SIZE = 24 + (2 * _BT_CAP + 1) * 8 + 2 * _BT_CAP * 8

    def _emit(self, w: _W) -> int:
        """Write this group (children first); returns object header addr."""
        child_addrs = {n: c._emit(w) for n, c in self.children.items()}
        # local heap: names (offset 0 reserved for the empty string)
        heap_data = bytearray(b"\x00" * 8)
        offs = {}
        names = sorted(child_addrs)
        for n in names:
            offs[n] = len(heap_data)
            nb = n.encode("utf-8") + b"\x00"
            heap_data += nb + b"\x00" * (_pad8(len(nb)) - len(nb))
        heap_seg = w.alloc(bytes(heap_data))
        heap = w.alloc(b"HEAP" + struct.pack("<BBBBQQQ", 0, 0, 0, 0,
                                             len(heap_data), 1, heap_seg))
        # symbol table nodes: chunks of <= 8 sorted entries
        snods: List[Tuple[int, str, str]] = []  # (addr, first, last)
        for i in range(0, len(names), self._SNOD_CAP):
            chunk = names[i:i + self._SNOD_CAP]
            snod = b"SNOD" + struct.pack("<BBH", 1, 0, len(chunk))
            for n in chunk:
                snod += struct.pack("<QQII16x", offs[n], child_addrs[n], 0, 0)
            snod += b"\x00" * (self._SNOD_SIZE - len(snod))
            snods.append((w.alloc(snod), chunk[0], chunk[-1]))

        def emit_bt(level: int, kids: List[Tuple[int, str, str]]) -> int:
            """One btree node over (addr, first_name, last_name) children."""
            node = b"TREE" + struct.pack("<BBHQQ", 0, level, len(kids),
                                         UNDEF, UNDEF)
            for i, (addr, first, last) in enumerate(kids):
                node += struct.pack("<Q", 0 if i == 0 else offs[first])
                node += struct.pack("<Q", addr)
            node += struct.pack("<Q", offs[kids[-1][2]] if kids else 0)
            node += b"\x00" * (self._BT_SIZE - len(node))
            return w.alloc(node)

        if not snods:
            btree_addr = emit_bt(0, [])
        else:
            level = 0
            nodes = snods
            while True:
                groups = [nodes[i:i + self._BT_CAP]
                          for i in range(0, len(nodes), self._BT_CAP)]
                layer = [(emit_bt(level, g), g[0][1], g[-1][2])
                         for g in groups]
                if len(layer) == 1:
                    btree_addr = layer[0][0]
                    break
                nodes = layer
                level += 1
        msgs = [_Msg(0x0011, struct.pack("<QQ", btree_addr, heap))]
        for an, av in self.attrs.items():
            msgs.append(_attr_msg(an, av))
        return w.alloc(_object_header(msgs))


class WDataset:
    def __init__(self, name: str, data: np.ndarray,
                 chunks: Optional[Tuple[int, ...]] = None):
        if data.dtype.kind not in "iufS":
            raise NotImplementedError(f"dataset dtype {data.dtype}")
        if chunks is not None:
            if len(chunks) != data.ndim:
                raise ValueError("chunks rank mismatch")
            if any(c < 1 for c in chunks):
                raise ValueError("chunk dims must be >= 1")
        self.name = name
        self.data = data
        self.chunks = tuple(chunks) if chunks else None
        self.attrs: Dict[str, object] = {}

    # chunk B-tree v1 (type 1): superblock v0 implies indexed-storage K=32,
    # so nodes hold up to 64 entries and readers fetch full-capacity images
    _CHUNK_BT_CAP = 64

    def _emit_chunked(self, w: _W) -> bytes:
        """Write chunk data + B-tree; returns the layout message body."""
        d, ck = self.data, self.chunks
        nd = d.ndim
        keysize = 8 + 8 * (nd + 1)
        node_size = 24 + (2 * self._CHUNK_BT_CAP + 1) * keysize \
            + 2 * self._CHUNK_BT_CAP * 8
        grid = [range(0, d.shape[i], ck[i]) for i in range(nd)]
        import itertools
        entries = []  # (offsets tuple, addr, nbytes)
        for offs in itertools.product(*grid):
            sl = tuple(slice(o, min(o + ck[i], d.shape[i]))
                       for i, o in enumerate(offs))
            block = np.zeros(ck, d.dtype)
            csl = tuple(slice(0, sl[i].stop - sl[i].start) for i in range(nd))
            block[csl] = d[sl]
            raw = block.tobytes()
            entries.append((offs, w.alloc(raw), len(raw)))

        def key(offs) -> bytes:
            k = struct.pack("<II", 0, 0)  # size+mask only meaningful pre-child
            for o in offs:
                k += struct.pack("<Q", o)
            return k + struct.pack("<Q", 0)

        def entry_key(e) -> bytes:
            k = struct.pack("<II", e[2], 0)
            for o in e[0]:
                k += struct.pack("<Q", o)
            return k + struct.pack("<Q", 0)

        end_offs = tuple(
            (d.shape[i] + ck[i] - 1) // ck[i] * ck[i] for i in range(nd))

        def emit_nodes(level: int, kids):
            """kids: list of (first_offs, addr, first_key_bytes)."""
            out = []
            for i in range(0, len(kids), self._CHUNK_BT_CAP):
                grp = kids[i:i + self._CHUNK_BT_CAP]
                node = b"TREE" + struct.pack("<BBHQQ", 1, level, len(grp),
                                             UNDEF, UNDEF)
                for first_offs, addr, kb in grp:
                    node += kb + struct.pack("<Q", addr)
                # final key: next sibling's first offsets or one-past-end
                j = i + len(grp)
                nxt = kids[j][2] if j < len(kids) else key(end_offs)
                node += nxt
                node += b"\x00" * (node_size - len(node))
                out.append((grp[0][0], w.alloc(node), grp[0][2]))
            return out

        layer = [(e[0], e[1], entry_key(e)) for e in entries]
        level = 0
        while True:
            layer = emit_nodes(level, layer)
            if len(layer) == 1:
                btree_addr = layer[0][1]
                break
            level += 1
        body = struct.pack("<BBBQ", 3, 2, nd + 1, btree_addr)
        for c in ck:
            body += struct.pack("<I", c)
        body += struct.pack("<I", d.dtype.itemsize)
        return body

    def _emit(self, w: _W) -> int:
        if self.chunks:
            layout = self._emit_chunked(w)
        else:
            raw_addr = w.alloc(self.data.tobytes())
            layout = struct.pack("<BBQQ", 3, 1, raw_addr, self.data.nbytes)
        msgs = [
            _Msg(0x0001, _ds_msg(self.data.shape)),
            _Msg(0x0003, _dt_msg(self.data.dtype)),
            _Msg(0x0005, struct.pack("<BBBB", 2, 2, 2, 0)),  # fill: undefined
            _Msg(0x0008, layout),
        ]
        for an, av in self.attrs.items():
            msgs.append(_attr_msg(an, av))
        return w.alloc(_object_header(msgs))


class H5Writer:
    """h5py-compatible writer for the reference layout (context manager).

    Usage mirrors the h5py surface the reference uses::

        with H5Writer(path) as f:
            g = f.create_group("ctg1_0-999")
            g["positions"] = pos_arr
            g.attrs["contig"] = "ctg1"
    """

    def __init__(self, path: str):
        self.path = path
        self.root = WGroup(self, "/")

    def create_group(self, name: str) -> WGroup:
        return self.root.create_group(name)

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    def close(self) -> None:
        w = _W()
        w.alloc(b"\x00" * 96)  # reserve superblock space at offset 0
        root_addr = self.root._emit(w)
        eof = _pad8(w.off)
        sb = (_SIG
              + struct.pack("<BBBBBBBB", 0, 0, 0, 0, 0, 8, 8, 0)
              + struct.pack("<HHI", 4, 16, 0)
              + struct.pack("<QQQQ", 0, UNDEF, eof, UNDEF)
              + struct.pack("<QQII16x", 0, root_addr, 0, 0))
        image = b"".join(w.parts)
        with open(self.path, "wb") as fh:
            fh.write(sb + image[len(sb):])
            pad = eof - w.off
            if pad:
                fh.write(b"\x00" * pad)
