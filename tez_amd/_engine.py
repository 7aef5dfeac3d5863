"""ctypes binding over libtezsort.so — the PRODUCT compute path.

No CPU fallback exists here by design: if the HIP extension is missing or a
GPU is unavailable, calls raise immediately.  The CPU oracle
under /root/repo/oracle is test-only and is never imported from this package.
"""
import ctypes
import os

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB = os.path.join(_DIR, "libtezsort.so")

KEY_BYTES = 0
KEY_TEXT = 1
CMP_TEZBYTES = 0
CMP_TEXT = 1


class TzsConf(ctypes.Structure):
    _fields_ = [
        ("num_partitions", ctypes.c_int32),
        ("key_type", ctypes.c_int32),
        ("value_type", ctypes.c_int32),
        ("comparator", ctypes.c_int32),
        ("rle", ctypes.c_int32),
        ("send_empty_partition_details", ctypes.c_int32),
        ("io_sort_factor", ctypes.c_int32),
        ("final_merge_enabled", ctypes.c_int32),
        ("sort_buffer_bytes", ctypes.c_int64),
        ("device", ctypes.c_int32),
        ("world_size", ctypes.c_int32),
        ("rank", ctypes.c_int32),
        ("combiner", ctypes.c_int32),
        ("min_spills_for_combine", ctypes.c_int32),
        ("discard_spill_streams", ctypes.c_int32),
    ]


class TzsIndexRecord(ctypes.Structure):
    _fields_ = [("start_offset", ctypes.c_int64),
                ("raw_length", ctypes.c_int64),
                ("part_length", ctypes.c_int64)]


class TzsSegment(ctypes.Structure):
    _fields_ = [("d_data", ctypes.c_void_p), ("d_off", ctypes.c_void_p),
                ("d_klen", ctypes.c_void_p), ("n", ctypes.c_int64)]


class TzsCounters(ctypes.Structure):
    _fields_ = [(n, ctypes.c_int64) for n in
                ("output_records", "output_bytes", "output_bytes_with_overhead",
                 "spilled_records", "num_spills", "rle_applied")]


class TzsTimes(ctypes.Structure):
    _fields_ = [(n, ctypes.c_int64) for n in
                ("absorb_ns", "composite_ns", "sort_ns", "permute_ns", "emit_ns",
                 "crc_ns", "dominant_kernel_elems", "total_ns", "sort_passes",
                 "dominant_kernel_ns", "merge_ns")]


_lib = None


def lib():
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_LIB):
        raise RuntimeError(
            f"tez_amd native engine missing: {_LIB} not built. "
            "Run __graft_entry__.build() (hipcc --offload-arch=gfx950). "
            "There is no CPU fallback.")
    # Load torch (and its bundled libamdhip64, SONAME libamdhip64.so.7) FIRST:
    # if this .so pulls /opt/rocm's runtime in first, torch later loads a
    # second HSA runtime in-process and sees zero GPUs.  With torch resident,
    # the dynamic linker resolves our libamdhip64.so.7 dependency to torch's
    # already-loaded copy — one runtime for both.
    try:
        import torch  # noqa: F401
    except Exception:
        pass
    L = ctypes.CDLL(_LIB)
    c = ctypes
    L.tzs_last_error.restype = c.c_char_p
    L.tzs_conf_default.argtypes = [c.POINTER(TzsConf), c.c_int32]
    L.tzs_sorter_create.argtypes = [c.POINTER(TzsConf), c.POINTER(c.c_void_p)]
    L.tzs_sorter_write.argtypes = [c.c_void_p, c.c_void_p, c.c_int32, c.c_void_p,
                                   c.c_int32, c.c_int32]
    L.tzs_sorter_write_batch_device.argtypes = [c.c_void_p, c.c_void_p, c.c_void_p,
                                                c.c_void_p, c.c_void_p, c.c_int64]
    L.tzs_sorter_write_batch_device_adopt.argtypes = [c.c_void_p, c.c_void_p,
                                                      c.c_void_p, c.c_void_p,
                                                      c.c_void_p, c.c_int64]
    L.tzs_sorter_spill.argtypes = [c.c_void_p]
    L.tzs_sorter_add_sorted_segment.argtypes = [c.c_void_p, c.c_void_p, c.c_void_p,
                                                c.c_void_p, c.c_void_p, c.c_int64]
    L.tzs_sorter_flush.argtypes = [c.c_void_p]
    L.tzs_sorter_num_spills.argtypes = [c.c_void_p]
    L.tzs_sorter_output.argtypes = [c.c_void_p, c.POINTER(c.c_void_p),
                                    c.POINTER(c.c_int64), c.POINTER(TzsIndexRecord)]
    L.tzs_sorter_output_compressed.argtypes = [c.c_void_p, c.POINTER(c.c_void_p),
                                               c.POINTER(c.c_int64),
                                               c.POINTER(TzsIndexRecord)]
    L.tzs_sorter_spill_output.argtypes = [c.c_void_p, c.c_int32, c.POINTER(c.c_void_p),
                                          c.POINTER(c.c_int64), c.POINTER(TzsIndexRecord)]
    L.tzs_sorter_write_files.argtypes = [c.c_void_p, c.c_char_p, c.c_char_p]
    L.tzs_sorter_sorted_columnar.argtypes = [c.c_void_p, c.POINTER(c.c_void_p),
                                             c.POINTER(c.c_void_p),
                                             c.POINTER(c.c_void_p),
                                             c.POINTER(c.c_uint64),
                                             c.POINTER(c.c_uint64)]
    L.tzs_memcpy_d2d.argtypes = [c.c_void_p, c.c_void_p, c.c_uint64]
    L.tzs_merge_segments.argtypes = [c.POINTER(TzsConf), c.POINTER(TzsSegment),
                                     c.c_int32, c.POINTER(c.c_void_p),
                                     c.POINTER(c.c_int64), c.POINTER(TzsIndexRecord)]
    L.tzs_sorter_counters.argtypes = [c.c_void_p, c.POINTER(TzsCounters)]
    L.tzs_sorter_times.argtypes = [c.c_void_p, c.POINTER(TzsTimes)]
    L.tzs_sorter_close.argtypes = [c.c_void_p]
    L.tzs_generate.argtypes = [c.c_uint64, c.c_int64, c.c_int32, c.c_int32, c.c_int32,
                               c.POINTER(TzsConf), c.POINTER(c.c_void_p),
                               c.POINTER(c.c_void_p), c.POINTER(c.c_void_p),
                               c.POINTER(c.c_void_p)]
    L.tzs_free_device.argtypes = [c.c_void_p]
    L.tzs_memcpy_d2h.argtypes = [c.c_void_p, c.c_void_p, c.c_uint64]
    L.tzs_memcpy_h2d.argtypes = [c.c_void_p, c.c_void_p, c.c_uint64]
    L.tzs_malloc_device.argtypes = [c.c_uint64, c.POINTER(c.c_void_p)]
    L.tzs_pool_stats.argtypes = [c.POINTER(c.c_uint64)]
    L.tzs_pool_stats.restype = None
    L.tzs_device_available.restype = c.c_int
    L.tzs_test_crc_combine.restype = c.c_uint32
    L.tzs_test_crc_combine.argtypes = [c.c_uint32, c.c_uint32, c.c_uint64]
    L.tzs_test_crc32.restype = c.c_uint32
    L.tzs_test_crc32.argtypes = [c.c_uint32, c.c_void_p, c.c_uint64]
    _lib = L
    return L


def _ck(rc, what):
    if rc < 0:
        raise RuntimeError(f"{what} failed rc={rc}: {lib().tzs_last_error().decode()}")
    return rc


def make_conf(num_partitions, **kw):
    c = TzsConf()
    lib().tzs_conf_default(ctypes.byref(c), num_partitions)
    for k, v in kw.items():
        setattr(c, k, v)
    return c


class Sorter:
    """The ExternalSorter replacement behind OrderedPartitionedKVOutput
    (see include/tezsort.h citations)."""

    def __init__(self, conf: TzsConf):
        self.conf = conf
        h = ctypes.c_void_p()
        _ck(lib().tzs_sorter_create(ctypes.byref(conf), ctypes.byref(h)), "create")
        self.h = h

    def write(self, key: bytes, val: bytes, partition: int = -1):
        _ck(lib().tzs_sorter_write(self.h, key, len(key), val, len(val), partition),
            "write")

    def write_batch_device(self, d_data, d_off, d_klen, d_part, n):
        _ck(lib().tzs_sorter_write_batch_device(
            self.h, d_data, d_off, d_klen, d_part, n), "write_batch")

    def write_batch_device_adopt(self, d_data, d_off, d_klen, d_part, n):
        """Zero-copy absorb: sorter takes ownership of tzs-allocated device
        buffers (first batch of a spill only); caller must not free them."""
        _ck(lib().tzs_sorter_write_batch_device_adopt(
            self.h, d_data, d_off, d_klen, d_part, n), "write_batch_adopt")

    def spill(self):
        return _ck(lib().tzs_sorter_spill(self.h), "spill")

    def add_sorted_segment(self, d_data, d_off, d_klen, d_part, n):
        """Ingest an already-sorted columnar segment without copy or re-sort
        (reduce-side MergeManager admission).  Caller keeps the device
        buffers alive until flush()/close()."""
        return _ck(lib().tzs_sorter_add_sorted_segment(
            self.h, d_data, d_off, d_klen, d_part, n), "add_sorted_segment")

    def flush(self):
        _ck(lib().tzs_sorter_flush(self.h), "flush")

    def num_spills(self):
        return lib().tzs_sorter_num_spills(self.h)

    def output(self):
        """Returns (ifile_bytes: bytes, index: list[(start,raw,part)])."""
        p = ctypes.c_void_p()
        n = ctypes.c_int64()
        idx = (TzsIndexRecord * self.conf.num_partitions)()
        _ck(lib().tzs_sorter_output(self.h, ctypes.byref(p), ctypes.byref(n), idx),
            "output")
        buf = bytearray(n.value)
        if n.value:
            ba = (ctypes.c_char * n.value).from_buffer(buf)
            _ck(lib().tzs_memcpy_d2h(ctypes.addressof(ba), p, n.value), "d2h")
        return bytes(buf), [(r.start_offset, r.raw_length, r.part_length) for r in idx]

    def output_compressed(self):
        """TIF\\1 compressed final stream (device deflate) + index."""
        p = ctypes.c_void_p()
        n = ctypes.c_int64()
        idx = (TzsIndexRecord * self.conf.num_partitions)()
        _ck(lib().tzs_sorter_output_compressed(self.h, ctypes.byref(p),
                                               ctypes.byref(n), idx),
            "output_compressed")
        buf = bytearray(n.value)
        if n.value:
            ba = (ctypes.c_char * n.value).from_buffer(buf)
            _ck(lib().tzs_memcpy_d2h(ctypes.addressof(ba), p, n.value), "d2h")
        return bytes(buf), [(r.start_offset, r.raw_length, r.part_length)
                            for r in idx]

    def spill_output(self, spill_id):
        p = ctypes.c_void_p()
        n = ctypes.c_int64()
        idx = (TzsIndexRecord * self.conf.num_partitions)()
        _ck(lib().tzs_sorter_spill_output(self.h, spill_id, ctypes.byref(p),
                                          ctypes.byref(n), idx), "spill_output")
        buf = bytearray(n.value)
        if n.value:
            ba = (ctypes.c_char * n.value).from_buffer(buf)
            _ck(lib().tzs_memcpy_d2h(ctypes.addressof(ba), p, n.value), "d2h")
        return bytes(buf), [(r.start_offset, r.raw_length, r.part_length) for r in idx]

    def output_meta(self):
        """(device_ptr, nbytes, index) without copying the stream to host."""
        p = ctypes.c_void_p()
        n = ctypes.c_int64()
        idx = (TzsIndexRecord * self.conf.num_partitions)()
        _ck(lib().tzs_sorter_output(self.h, ctypes.byref(p), ctypes.byref(n), idx),
            "output")
        return p, n.value, [(r.start_offset, r.raw_length, r.part_length) for r in idx]

    def sorted_columnar(self):
        """Final sort as device columnar arrays + partition ranges
        (the exchange wire — DESIGN.md §4).  Returns
        (d_data, d_off, d_klen, rec_ranges: list, byte_ranges: list)."""
        P = self.conf.num_partitions
        d = ctypes.c_void_p()
        o = ctypes.c_void_p()
        k = ctypes.c_void_p()
        rr = (ctypes.c_uint64 * (P + 1))()
        br = (ctypes.c_uint64 * (P + 1))()
        _ck(lib().tzs_sorter_sorted_columnar(self.h, ctypes.byref(d), ctypes.byref(o),
                                             ctypes.byref(k), rr, br),
            "sorted_columnar")
        return d, o, k, list(rr), list(br)

    def write_files_compressed(self, local_dir: str, unique_id: str):
        """Reference on-disk layout with DefaultCodec segments: the TIF\\1
        stream from the device deflate as file.out + the matching index
        (partLength = compressed) — servable by the ShuffleHandler and
        consumable by stock fetchers (IFile.java:352-368)."""
        import os
        import zlib
        data, index = self.output_compressed()
        base = os.path.join(local_dir, "output", unique_id)
        os.makedirs(base, exist_ok=True)
        with open(os.path.join(base, "file.out"), "wb") as f:
            f.write(data)
        ix = bytearray()
        for st, raw, cl in index:
            ix += int(st).to_bytes(8, "big") + int(raw).to_bytes(8, "big") \
                + int(cl).to_bytes(8, "big")
        ix += zlib.crc32(bytes(ix)).to_bytes(8, "big")
        with open(os.path.join(base, "file.out.index"), "wb") as f:
            f.write(bytes(ix))

    def write_files(self, local_dir: str, unique_id: str):
        _ck(lib().tzs_sorter_write_files(self.h, local_dir.encode(), unique_id.encode()),
            "write_files")

    def counters(self):
        c = TzsCounters()
        _ck(lib().tzs_sorter_counters(self.h, ctypes.byref(c)), "counters")
        return {n: getattr(c, n) for n, _ in c._fields_}

    def times(self):
        t = TzsTimes()
        _ck(lib().tzs_sorter_times(self.h, ctypes.byref(t)), "times")
        return {n: getattr(t, n) for n, _ in t._fields_}

    def close(self):
        if self.h:
            lib().tzs_sorter_close(self.h)
            self.h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


def upload_records(data: bytes, off, klen, part=None):
    """Upload host columnar records to the device; returns device pointers
    (caller frees via free_device).  off: list/array u64 [n+1]; klen u32 [n]."""
    import numpy as np
    L = lib()
    n = len(klen)
    d = ctypes.c_void_p()
    o = ctypes.c_void_p()
    k = ctypes.c_void_p()
    p = ctypes.c_void_p()
    data_n = max(len(data), 1)
    _ck(L.tzs_malloc_device(data_n, ctypes.byref(d)), "malloc")
    _ck(L.tzs_malloc_device(8 * (n + 1), ctypes.byref(o)), "malloc")
    _ck(L.tzs_malloc_device(max(4 * n, 4), ctypes.byref(k)), "malloc")
    if data:
        _ck(L.tzs_memcpy_h2d(d, data, len(data)), "h2d")
    off_a = np.asarray(off, dtype=np.uint64)
    klen_a = np.asarray(klen, dtype=np.uint32)
    _ck(L.tzs_memcpy_h2d(o, off_a.ctypes.data, 8 * (n + 1)), "h2d")
    if n:
        _ck(L.tzs_memcpy_h2d(k, klen_a.ctypes.data, 4 * n), "h2d")
    if part is not None:
        part_a = np.asarray(part, dtype=np.int32)
        _ck(L.tzs_malloc_device(max(4 * n, 4), ctypes.byref(p)), "malloc")
        if n:
            _ck(L.tzs_memcpy_h2d(p, part_a.ctypes.data, 4 * n), "h2d")
        return d, o, k, p
    return d, o, k, None


def generate(seed, n, kind, klen, vlen, conf):
    L = lib()
    d = ctypes.c_void_p()
    o = ctypes.c_void_p()
    k = ctypes.c_void_p()
    p = ctypes.c_void_p()
    _ck(L.tzs_generate(seed, n, kind, klen, vlen, ctypes.byref(conf), ctypes.byref(d),
                       ctypes.byref(o), ctypes.byref(k), ctypes.byref(p)), "generate")
    return d, o, k, p


def free_device(*ptrs):
    for p in ptrs:
        if p:
            lib().tzs_free_device(p)


def read_device(dev_ptr, offset, nbytes):
    """Copy [offset, offset+nbytes) of a device buffer to host bytes."""
    buf = bytearray(nbytes)
    if nbytes:
        ba = (ctypes.c_char * nbytes).from_buffer(buf)
        _ck(lib().tzs_memcpy_d2h(ctypes.addressof(ba),
                                 ctypes.c_void_p((dev_ptr.value if hasattr(dev_ptr, "value")
                                                  else dev_ptr) + offset), nbytes), "d2h")
    return bytes(buf)


def merge_segments(conf, segments):
    """tzs_merge_segments: segments = list of (d_data, d_off, d_klen, n)
    device-pointer tuples (one partition).  Returns (bytes, (start,raw,part))."""
    n = len(segments)
    arr = (TzsSegment * max(n, 1))()
    for i, (d, o, k, cnt) in enumerate(segments):
        arr[i].d_data = d if isinstance(d, int) else d.value
        arr[i].d_off = o if isinstance(o, int) else o.value
        arr[i].d_klen = k if isinstance(k, int) else k.value
        arr[i].n = cnt
    out = ctypes.c_void_p()
    ln = ctypes.c_int64()
    rec = TzsIndexRecord()
    _ck(lib().tzs_merge_segments(ctypes.byref(conf), arr, n, ctypes.byref(out),
                                 ctypes.byref(ln), ctypes.byref(rec)), "merge_segments")
    data = read_device(out, 0, ln.value)
    if out.value:
        lib().tzs_free_device(out)
    return data, (rec.start_offset, rec.raw_length, rec.part_length)


def pool_stats():
    """{inuse, held, peak, drops} of the device buffer pool (bytes/counts)."""
    arr = (ctypes.c_uint64 * 4)()
    lib().tzs_pool_stats(arr)
    return {"inuse": arr[0], "held": arr[1], "peak": arr[2], "drops": arr[3]}


def device_available():
    try:
        return bool(lib().tzs_device_available())
    except RuntimeError:
        return False
