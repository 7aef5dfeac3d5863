"""ORACLE — ctypes wrapper over liboracle.so (CPU restatement of the tez
ordered-shuffle hot path). TEST INFRASTRUCTURE ONLY: imported by tests/,
__graft_entry__.smoke() (as the checker) and bench.py's cpu_baseline leg.
The product package (tez_amd) must never import this. See oracle/tzoracle.c
for the reference citations."""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_DIR, "liboracle.so")


def _build():
    subprocess.run(["make", "-C", _DIR, "-s"], check=True)


def _load():
    if not os.path.exists(_LIB_PATH) or os.path.getmtime(_LIB_PATH) < os.path.getmtime(
        os.path.join(_DIR, "tzoracle.c")
    ):
        _build()
    lib = ctypes.CDLL(_LIB_PATH)
    c = ctypes
    u8p = c.POINTER(c.c_uint8)
    lib.tzo_vint_size.restype = c.c_int
    lib.tzo_vint_size.argtypes = [c.c_int64]
    lib.tzo_vint_write.restype = c.c_int
    lib.tzo_vint_write.argtypes = [u8p, c.c_int64]
    lib.tzo_vint_read.restype = c.c_int
    lib.tzo_vint_read.argtypes = [u8p, c.POINTER(c.c_int64)]
    lib.tzo_crc32.restype = c.c_uint32
    lib.tzo_crc32.argtypes = [c.c_uint32, u8p, c.c_size_t]
    lib.tzo_hash_bytes.restype = c.c_int32
    lib.tzo_hash_bytes.argtypes = [u8p, c.c_int32]
    lib.tzo_partition.restype = c.c_int32
    lib.tzo_partition.argtypes = [u8p, c.c_int32, c.c_int32]
    lib.tzo_prefix.restype = c.c_uint32
    lib.tzo_prefix.argtypes = [c.c_int, c.c_int, c.c_int32, c.c_int32, u8p, c.c_int32]
    lib.tzo_compare_key.restype = c.c_int
    lib.tzo_compare_key.argtypes = [c.c_int, u8p, c.c_int32, u8p, c.c_int32]
    lib.tzo_writer_new.restype = c.c_void_p
    lib.tzo_writer_new.argtypes = [c.c_int]
    lib.tzo_writer_append.restype = c.c_int
    lib.tzo_writer_append.argtypes = [c.c_void_p, u8p, c.c_int32, u8p, c.c_int32]
    lib.tzo_writer_append_same.restype = c.c_int
    lib.tzo_writer_append_same.argtypes = [c.c_void_p, u8p, c.c_int32]
    lib.tzo_writer_close.restype = c.c_int
    lib.tzo_writer_close.argtypes = [
        c.c_void_p, c.POINTER(c.c_void_p), c.POINTER(c.c_int64),
        c.POINTER(c.c_int64), c.POINTER(c.c_int64)]
    lib.tzo_writer_free.argtypes = [c.c_void_p]
    lib.tzo_ifile_read.restype = c.c_int
    lib.tzo_ifile_read.argtypes = [u8p, c.c_int64, c.c_int, c.POINTER(c.c_void_p)]
    for name, res in [("tzo_records_n", c.c_int64),
                      ("tzo_records_keys", c.POINTER(c.c_uint8)),
                      ("tzo_records_key_off", c.POINTER(c.c_int64)),
                      ("tzo_records_vals", c.POINTER(c.c_uint8)),
                      ("tzo_records_val_off", c.POINTER(c.c_int64)),
                      ("tzo_records_same", c.POINTER(c.c_uint8))]:
        getattr(lib, name).restype = res
        getattr(lib, name).argtypes = [c.c_void_p]
    lib.tzo_records_free.argtypes = [c.c_void_p]
    lib.tzo_index_encode.restype = c.c_int
    lib.tzo_index_encode.argtypes = [c.POINTER(c.c_int64), c.c_int32, u8p]
    lib.tzo_spill.restype = c.c_int
    lib.tzo_spill.argtypes = [
        u8p, c.POINTER(c.c_uint64), c.POINTER(c.c_uint32), c.POINTER(c.c_int32),
        c.c_int64, c.c_int32, c.c_int, c.c_int, c.c_int, c.c_int, c.c_int,
        c.POINTER(c.c_void_p), c.POINTER(c.c_int64),
        c.POINTER(c.c_void_p), c.POINTER(c.c_int64),
        c.POINTER(c.c_int64), c.POINTER(c.c_int)]
    lib.tzo_spill_mt.restype = c.c_int
    lib.tzo_spill_mt.argtypes = [
        u8p, c.POINTER(c.c_uint64), c.POINTER(c.c_uint32), c.POINTER(c.c_int32),
        c.c_int64, c.c_int32, c.c_int, c.c_int, c.c_int, c.c_int, c.c_int,
        c.POINTER(c.c_void_p), c.POINTER(c.c_int64),
        c.POINTER(c.c_void_p), c.POINTER(c.c_int64)]
    lib.tzo_final_merge.restype = c.c_int
    lib.tzo_final_merge.argtypes = [
        c.POINTER(c.c_void_p), c.POINTER(c.c_int64), c.POINTER(c.c_void_p),
        c.c_int32, c.c_int32, c.c_int, c.c_int, c.c_int, c.c_int32, c.c_int,
        c.POINTER(c.c_void_p), c.POINTER(c.c_int64),
        c.POINTER(c.c_void_p), c.POINTER(c.c_int64)]
    lib.tzo_free.argtypes = [c.c_void_p]
    lib.tzo_shuffle_header_write.restype = c.c_int
    lib.tzo_shuffle_header_write.argtypes = [u8p, c.c_char_p, c.c_int64, c.c_int64, c.c_int32]
    lib.tzo_shuffle_header_read.restype = c.c_int
    lib.tzo_shuffle_header_read.argtypes = [
        u8p, c.c_char_p, c.c_int, c.POINTER(c.c_int64), c.POINTER(c.c_int64),
        c.POINTER(c.c_int32)]
    return lib


_lib = _load()

KEY_BYTES = 0
KEY_TEXT = 1
CMP_TEZBYTES = 0
CMP_TEXT = 1


def _u8p(arr):
    return arr.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8))


def _take_buf(pp, ln):
    """Copy a malloc'd buffer into bytes and free it."""
    if not pp.value or ln.value == 0:
        if pp.value:
            _lib.tzo_free(pp.value)
        return b""
    out = ctypes.string_at(pp.value, ln.value)
    _lib.tzo_free(pp.value)
    return out


def vint_encode(v):
    buf = np.zeros(10, dtype=np.uint8)
    n = _lib.tzo_vint_write(_u8p(buf), v)
    return buf[:n].tobytes()


def vint_decode(b, off=0):
    arr = np.frombuffer(b, dtype=np.uint8)[off:].copy()
    out = ctypes.c_int64()
    n = _lib.tzo_vint_read(_u8p(arr), ctypes.byref(out))
    return out.value, n


def crc32(data, crc=0):
    arr = np.frombuffer(data, dtype=np.uint8).copy()
    return _lib.tzo_crc32(crc, _u8p(arr), arr.size)


def hash_bytes(data):
    arr = np.frombuffer(data, dtype=np.uint8).copy()
    return _lib.tzo_hash_bytes(_u8p(arr), arr.size)


def partition_of(content, nparts):
    arr = np.frombuffer(content, dtype=np.uint8).copy()
    return _lib.tzo_partition(_u8p(arr), arr.size, nparts)


def serialize_bytes_writable(content: bytes) -> bytes:
    """hadoop BytesWritable.write: 4B BE length + content."""
    return len(content).to_bytes(4, "big") + content


def serialize_text(content: bytes) -> bytes:
    """hadoop Text.write: vint(len) + UTF-8 bytes."""
    return vint_encode(len(content)) + content


def build_records(serialized_pairs):
    """serialized_pairs: list of (ser_key: bytes, ser_val: bytes).
    Returns (data: np.uint8, off: np.uint64 [n+1], klen: np.uint32 [n])."""
    n = len(serialized_pairs)
    off = np.zeros(n + 1, dtype=np.uint64)
    klen = np.zeros(n, dtype=np.uint32)
    parts = []
    pos = 0
    for i, (k, v) in enumerate(serialized_pairs):
        parts.append(k)
        parts.append(v)
        klen[i] = len(k)
        pos += len(k) + len(v)
        off[i + 1] = pos
    data = np.frombuffer(b"".join(parts), dtype=np.uint8).copy()
    if data.size == 0:
        data = np.zeros(1, dtype=np.uint8)
    return data, off, klen


def spill(data, off, klen, num_partitions, key_type=KEY_BYTES,
          comparator=CMP_TEZBYTES, rle_mode=-1, send_empty=True,
          partitions=None, want_order=False, combiner=0):
    """One PipelinedSorter spill (see tzo_spill). Returns dict with
    data/index bytes, rle flag, and optionally the sorted order."""
    n = len(klen)
    od = ctypes.c_void_p()
    odl = ctypes.c_int64()
    oi = ctypes.c_void_p()
    oil = ctypes.c_int64()
    rle = ctypes.c_int()
    order = np.zeros(max(n, 1), dtype=np.int64)
    pp = partitions.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)) if partitions is not None else None
    rc = _lib.tzo_spill(
        _u8p(data), off.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        klen.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)), pp,
        n, num_partitions, key_type, comparator, rle_mode, int(send_empty),
        combiner,
        ctypes.byref(od), ctypes.byref(odl), ctypes.byref(oi), ctypes.byref(oil),
        order.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)) if want_order else None,
        ctypes.byref(rle))
    assert rc == 0, f"tzo_spill rc={rc}"
    res = {"data": _take_buf(od, odl), "index": _take_buf(oi, oil), "rle": rle.value}
    if want_order:
        res["order"] = order[:n]
    return res


def final_merge(spills, num_partitions, comparator=CMP_TEZBYTES, rle_mode=-1,
                send_empty=True, factor=100, combiner=0):
    """PipelinedSorter.flush final merge over spill outputs
    (list of dicts with 'data'/'index'). numSpills==1 is a rename
    (PipelinedSorter.java:731-757): returns the spill unchanged."""
    if len(spills) == 1:
        return {"data": spills[0]["data"], "index": spills[0]["index"]}
    n = len(spills)
    datas = [np.frombuffer(s["data"], dtype=np.uint8).copy() for s in spills]
    for d in datas:
        if d.size == 0:
            d.resize(1, refcheck=False)
    idxs = [np.frombuffer(s["index"], dtype=np.uint8).copy() for s in spills]
    dptr = (ctypes.c_void_p * n)(*[d.ctypes.data for d in datas])
    dlen = (ctypes.c_int64 * n)(*[len(s["data"]) for s in spills])
    iptr = (ctypes.c_void_p * n)(*[i.ctypes.data for i in idxs])
    od = ctypes.c_void_p()
    odl = ctypes.c_int64()
    oi = ctypes.c_void_p()
    oil = ctypes.c_int64()
    rc = _lib.tzo_final_merge(dptr, dlen, iptr, n, num_partitions, comparator,
                              rle_mode, int(send_empty), factor, combiner,
                              ctypes.byref(od), ctypes.byref(odl),
                              ctypes.byref(oi), ctypes.byref(oil))
    assert rc == 0, f"tzo_final_merge rc={rc}"
    return {"data": _take_buf(od, odl), "index": _take_buf(oi, oil)}


def ifile_read(stream, with_header=True):
    """Parse an IFile stream; returns list of (key, value, same_key)."""
    arr = np.frombuffer(stream, dtype=np.uint8).copy()
    if arr.size == 0:
        arr = np.zeros(1, dtype=np.uint8)
    h = ctypes.c_void_p()
    rc = _lib.tzo_ifile_read(_u8p(arr), len(stream), int(with_header), ctypes.byref(h))
    if rc != 0:
        raise ValueError(f"ifile_read rc={rc}")
    n = _lib.tzo_records_n(h)
    keys = _lib.tzo_records_keys(h)
    koff = _lib.tzo_records_key_off(h)
    vals = _lib.tzo_records_vals(h)
    voff = _lib.tzo_records_val_off(h)
    same = _lib.tzo_records_same(h)
    out = []
    for i in range(n):
        k = ctypes.string_at(ctypes.addressof(keys.contents) + koff[i], koff[i + 1] - koff[i]) if koff[i + 1] > koff[i] else b""
        v = ctypes.string_at(ctypes.addressof(vals.contents) + voff[i], voff[i + 1] - voff[i]) if voff[i + 1] > voff[i] else b""
        out.append((k, v, bool(same[i])))
    _lib.tzo_records_free(h)
    return out


def index_decode(index_bytes, num_partitions, verify=True):
    """Parse a spill index file: list of (start, raw, part)."""
    assert len(index_bytes) == 24 * num_partitions + 8
    body = index_bytes[: 24 * num_partitions]
    if verify:
        want = int.from_bytes(index_bytes[-8:], "big")
        got = crc32(body)
        assert want == got, "index CRC mismatch"
    out = []
    for p in range(num_partitions):
        s = int.from_bytes(body[24 * p: 24 * p + 8], "big")
        r = int.from_bytes(body[24 * p + 8: 24 * p + 16], "big")
        c = int.from_bytes(body[24 * p + 16: 24 * p + 24], "big")
        out.append((s, r, c))
    return out


def shuffle_header_encode(map_id: str, clen: int, rlen: int, partition: int) -> bytes:
    buf = np.zeros(64 + len(map_id), dtype=np.uint8)
    n = _lib.tzo_shuffle_header_write(_u8p(buf), map_id.encode(), clen, rlen, partition)
    return buf[:n].tobytes()


def shuffle_header_decode(b: bytes):
    arr = np.frombuffer(b, dtype=np.uint8).copy()
    mid = ctypes.create_string_buffer(2048)
    clen = ctypes.c_int64()
    rlen = ctypes.c_int64()
    part = ctypes.c_int32()
    n = _lib.tzo_shuffle_header_read(_u8p(arr), mid, 2048, ctypes.byref(clen),
                                     ctypes.byref(rlen), ctypes.byref(part))
    assert n > 0
    return mid.value.decode(), clen.value, rlen.value, part.value, n


def spill_mt(data, off, klen, num_partitions, nthreads, key_type=KEY_BYTES,
             comparator=CMP_TEZBYTES, rle_mode=0, send_empty=True,
             partitions=None):
    """Partition-parallel tzo_spill (the multi-core CPU baseline,
    BASELINE.md). Byte-identical to spill() for unique-key inputs."""
    n = len(klen)
    od = ctypes.c_void_p()
    odl = ctypes.c_int64()
    oi = ctypes.c_void_p()
    oil = ctypes.c_int64()
    pp = partitions.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)) if partitions is not None else None
    rc = _lib.tzo_spill_mt(
        _u8p(data), off.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        klen.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)), pp,
        n, num_partitions, key_type, comparator, rle_mode, int(send_empty),
        nthreads,
        ctypes.byref(od), ctypes.byref(odl), ctypes.byref(oi), ctypes.byref(oil))
    assert rc == 0, f"tzo_spill_mt rc={rc}"
    return {"data": _take_buf(od, odl), "index": _take_buf(oi, oil)}
