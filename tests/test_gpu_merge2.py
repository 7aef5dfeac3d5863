"""Round-2 GPU merge coverage:

- merge-path k-way merge over retained sorted spills is bit-exact vs the
  oracle's TezMerger heap merge, including >32 segments (no kernarg bound,
  no coalesce copy), the C4-like 199-segment shape, and mixed
  uniform/variable-length spills (serialized-composite rebuild);
- explicit-partition multi-spill merges (the round-1 rc=-22 refusal);
- tzs_sorter_add_sorted_segment (reduce-side pre-sorted ingestion);
- mid-size (>=1e6) bit-exact parity through the size-gated branches
  (dense level keys at m >= n/2, upfront pass-skip histograms);
- the documented (partition, truncated proxy, serialized) total order for
  variable-length TezBytes multi-spill merges.
"""
import random

import numpy as np
import pytest

import oracle as o

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def engine():
    import __graft_entry__
    __graft_entry__.build()
    import tez_amd
    if not tez_amd.device_available():
        pytest.skip("no GPU")
    return tez_amd


def _mk_fixed(n, klen, vlen, seed):
    rng = random.Random(seed)
    seen = set()
    pairs = []
    while len(pairs) < n:
        k = bytes(rng.randrange(256) for _ in range(klen))
        if k in seen:
            continue
        seen.add(k)
        v = bytes(rng.randrange(256) for _ in range(vlen))
        pairs.append((o.serialize_bytes_writable(k), o.serialize_bytes_writable(v)))
    return pairs


def test_map_merge_199_spills(engine):
    """199 map-side spills merged at flush (> the old 32-spill kernarg bound,
    C4's per-reducer segment count): byte-exact vs the oracle final merge
    (TezMerger.java:921-931 handles any k via pass factor; unique keys make
    the multipass restatement byte-identical to one pass)."""
    P = 8
    pairs = _mk_fixed(199 * 60, 10, 12, seed=41)
    conf = engine.make_conf(P, key_type=engine.KEY_BYTES,
                            comparator=engine.CMP_TEZBYTES)
    s = engine.Sorter(conf)
    spills = []
    for i in range(199):
        batch = pairs[i * 60:(i + 1) * 60]
        for k, v in batch:
            s.write(k, v, -1)
        s.spill()
        d, f, kl = o.build_records(batch)
        spills.append(o.spill(d, f, kl, P))
    assert s.num_spills() == 199
    s.flush()
    got, gidx = s.output()
    s.close()
    want = o.final_merge(spills, P)
    assert gidx == o.index_decode(want["index"], P)
    assert got == want["data"]


def test_explicit_partition_multi_spill(engine):
    """TotalOrderPartitioner-style explicit partitions across MULTIPLE spills
    (C5 with forced spills; round 1 refused this with rc=-22).  Partition
    ids ride the retained composite top bits through the merge."""
    P = 16
    rng = random.Random(7)
    batches = []
    for b in range(5):
        batch = []
        for i in range(800):
            k = bytes([rng.randrange(256) for _ in range(10)])
            v = bytes([rng.randrange(256) for _ in range(20)])
            # range partition on the leading byte (TotalOrderPartitioner-ish)
            part = k[0] * P // 256
            batch.append((o.serialize_bytes_writable(k),
                          o.serialize_bytes_writable(v), part))
        batches.append(batch)
    conf = engine.make_conf(P, key_type=engine.KEY_BYTES,
                            comparator=engine.CMP_TEZBYTES)
    s = engine.Sorter(conf)
    spills = []
    for batch in batches:
        for k, v, part in batch:
            s.write(k, v, part)
        s.spill()
        d, f, kl = o.build_records([(k, v) for k, v, _ in batch])
        parts = np.array([p for _, _, p in batch], dtype=np.int32)
        spills.append(o.spill(d, f, kl, P, partitions=parts))
    s.flush()
    got, gidx = s.output()
    s.close()
    want = o.final_merge(spills, P)
    assert gidx == o.index_decode(want["index"], P)
    assert got == want["data"]


def test_mixed_uniform_variable_spills(engine):
    """One uniform-klen spill (content-form composites) + one variable-length
    spill (serialized-form): the merge rebuilds the content-form spill's
    retained composites in serialized form (DESIGN.md par.3 equivalence for
    uniform klen) and pins the documented total order."""
    P = 4
    rng = random.Random(19)
    uni = []
    seen = set()
    while len(uni) < 1200:
        k = bytes(rng.randrange(256) for _ in range(8))
        if k in seen:
            continue
        seen.add(k)
        uni.append((o.serialize_bytes_writable(k),
                    o.serialize_bytes_writable(b"u%04d" % len(uni))))
    var = []
    while len(var) < 1200:
        k = bytes(rng.randrange(256) for _ in range(rng.randrange(1, 13)))
        if k in seen:
            continue
        seen.add(k)
        var.append((o.serialize_bytes_writable(k),
                    o.serialize_bytes_writable(b"v%04d" % len(var))))
    conf = engine.make_conf(P, key_type=engine.KEY_BYTES,
                            comparator=engine.CMP_TEZBYTES)
    s = engine.Sorter(conf)
    for batch in (uni, var):
        for k, v in batch:
            s.write(k, v, -1)
        s.spill()
    s.flush()
    got, gidx = s.output()
    s.close()
    # variable-length TezBytes multi-spill merges pin the engine's documented
    # total order (the reference's own output is emergent here — DESIGN.md
    # par.3); single-spill/uniform shapes stay oracle-bit-exact elsewhere
    assert _read_rows(got, gidx, P) == _expect_documented(uni + var, P)


def _doc_key(kser, P):
    """The engine's documented TezBytes order within a partition:
    (truncated proxy, serialized bytes) — DESIGN.md par.3."""
    ref_pb = 1
    v2 = P
    while v2:
        ref_pb += 1
        v2 >>= 1
    pw = max(24 - ref_pb, 0)
    content = kser[4:]
    proxy = ((content[0] if len(content) > 0 else 0) << 16) \
        | ((content[1] if len(content) > 1 else 0) << 8) \
        | (content[2] if len(content) > 2 else 0)
    return ((proxy >> (24 - pw)) if pw else 0, kser)


def _expect_documented(pool, P):
    by_part = {}
    for k, v in pool:
        part = (_java_hash(k[4:]) & 0x7FFFFFFF) % P
        by_part.setdefault(part, []).append((k, v))
    expect = []
    for p in range(P):
        expect.extend(sorted(by_part.get(p, []),
                             key=lambda kv: _doc_key(kv[0], P)))
    return expect


def _read_rows(got, gidx, P):
    rows = []
    for p in range(P):
        st, raw, cl = gidx[p]
        if cl:
            for k, v, _same in o.ifile_read(got[st:st + cl], with_header=True):
                rows.append((k, v))
    return rows


def _java_hash(content):
    h = 1
    for b in content:
        sb = b - 256 if b >= 128 else b
        h = (31 * h + sb) & 0xFFFFFFFF
    if h >= 0x80000000:
        h -= 0x100000000
    return h


def _tezbytes_sorted(pairs, P):
    """Sort serialized (k, v) pairs the engine's documented way:
    (hash partition, truncated proxy, serialized key bytes) — the order the
    map side hands to the exchange."""
    def keyf(kv):
        k, _ = kv
        content = k[4:]
        part = (_java_hash(content) & 0x7FFFFFFF) % P
        return (part,) + _doc_key(k, P)
    return sorted(pairs, key=keyf)


def test_add_sorted_segment_199(engine):
    """Reduce-side pre-sorted ingestion at C4's 199-segment fan-in: each
    segment is one source's sorted columnar chunk; flush k-way merges them.
    Unique keys => byte-exact vs a plain oracle spill of the union."""
    P = 3
    pairs = _mk_fixed(199 * 50, 12, 16, seed=53)
    conf = engine.make_conf(P, key_type=engine.KEY_BYTES,
                            comparator=engine.CMP_TEZBYTES)
    s = engine.Sorter(conf)
    keep = []
    for i in range(199):
        seg = _tezbytes_sorted(pairs[i * 50:(i + 1) * 50], P)
        d, f, kl = o.build_records(seg)
        dd, ofp, kp, _ = engine.upload_records(d.tobytes(), f, kl)
        keep.append((dd, ofp, kp))
        s.add_sorted_segment(dd, ofp, kp, None, len(seg))
    s.flush()
    got, gidx = s.output()
    s.close()
    du, fu, klu = o.build_records(pairs)
    want = o.spill(du, fu, klu, P)
    assert gidx == o.index_decode(want["index"], P)
    assert got == want["data"]
    for bufs in keep:
        engine.free_device(*bufs)


def test_add_sorted_segment_cross_duplicates(engine):
    """Cross-segment duplicate keys through the pre-sorted reduce path: the
    SAME_KEY machine must emit RLE for equal keys meeting across segments
    (TezMerger.java:598-653) — byte-exact vs the oracle's heap merge over
    the same segments."""
    P = 2
    # all keys the same length: proxy order == serialized order, so the
    # engine's merge order equals the oracle's comparator order exactly
    shared = [o.serialize_bytes_writable(b"dup-key-%02d" % i) for i in range(6)]
    segs = []
    for sid in range(3):
        seg = []
        for i, k in enumerate(shared):
            seg.append((k, o.serialize_bytes_writable(b"s%dv%d" % (sid, i))))
        seg.append((o.serialize_bytes_writable(b"only-%05d" % sid),
                    o.serialize_bytes_writable(b"x")))
        segs.append(_tezbytes_sorted(seg, P))
    conf = engine.make_conf(P, key_type=engine.KEY_BYTES,
                            comparator=engine.CMP_TEZBYTES)
    s = engine.Sorter(conf)
    keep = []
    for seg in segs:
        d, f, kl = o.build_records(seg)
        dd, ofp, kp, _ = engine.upload_records(d.tobytes(), f, kl)
        keep.append((dd, ofp, kp))
        s.add_sorted_segment(dd, ofp, kp, None, len(seg))
    s.flush()
    got, gidx = s.output()
    s.close()
    # oracle: each segment spilled with rle OFF (raw columnar segments carry
    # no source RLE markers), then the TezMerger heap merge
    spills = []
    for seg in segs:
        d, f, kl = o.build_records(seg)
        spills.append(o.spill(d, f, kl, P, rle_mode=0))
    want = o.final_merge(spills, P)
    assert gidx == o.index_decode(want["index"], P)
    assert got == want["data"]
    for bufs in keep:
        engine.free_device(*bufs)


def test_add_sorted_segment_rejects_unsorted(engine):
    pairs = _mk_fixed(100, 8, 8, seed=77)
    d, f, kl = o.build_records(pairs)  # random order: not sorted
    dd, ofp, kp, _ = engine.upload_records(d.tobytes(), f, kl)
    conf = engine.make_conf(4)
    s = engine.Sorter(conf)
    with pytest.raises(RuntimeError, match="not sorted"):
        s.add_sorted_segment(dd, ofp, kp, None, len(pairs))
    s.close()
    engine.free_device(dd, ofp, kp)


def _upload_fixed_numpy(engine, n, klen, vlen, seed):
    rng = np.random.default_rng(seed)
    rec = 4 + klen + 4 + vlen
    view = np.zeros((n, rec), dtype=np.uint8)
    view[:, 0:4] = np.frombuffer(np.int32(klen).byteswap().tobytes(), dtype=np.uint8)
    view[:, 4:4 + klen] = rng.integers(0, 256, size=(n, klen), dtype=np.uint8)
    ids = np.arange(n, dtype=np.uint32).view(np.uint8).reshape(n, 4)
    view[:, 4:8] ^= ids  # force uniqueness
    view[:, 4 + klen:8 + klen] = np.frombuffer(
        np.int32(vlen).byteswap().tobytes(), dtype=np.uint8)
    view[:, 8 + klen:] = rng.integers(0, 256, size=(n, vlen), dtype=np.uint8)
    data = view.reshape(-1)
    off = np.arange(0, rec * (n + 1), rec, dtype=np.uint64)
    klens = np.full(n, 4 + klen, dtype=np.uint32)
    return data, off, klens


def test_midsize_bitexact_fixed(engine):
    """2e6-record bit-exact parity: large enough for
    the adaptive pass-count/onesweep gates (n >= 2e4, >= 1e5) while the
    oracle still runs in seconds."""
    n, P = 2_000_000, 64
    data, off, klens = _upload_fixed_numpy(engine, n, 16, 24, seed=6)
    dd, ofp, kp, _ = engine.upload_records(data.tobytes(), off, klens)
    conf = engine.make_conf(P)
    s = engine.Sorter(conf)
    s.write_batch_device(dd, ofp, kp, None, n)
    s.flush()
    got, gidx = s.output()
    s.close()
    engine.free_device(dd, ofp, kp)
    want = o.spill(data, off, klens, P)
    assert gidx == o.index_decode(want["index"], P)
    assert got == want["data"]


def test_midsize_bitexact_text_shared_prefix(engine):
    """1e6 Text keys sharing a 10-char prefix: nearly every record is
    ambiguous after the 8-byte composite, so m ~ n drives the dense
    level-key branch (m >= n/2), the in-run refinement machinery, and the
    upfront pass-skip histograms at refinement scale — bit-exact."""
    n, P = 1_000_000, 32
    rng = np.random.default_rng(11)
    suffix = rng.integers(ord('a'), ord('z') + 1, size=(n, 6), dtype=np.uint8)
    # unique suffixes: overwrite with base-26 record id
    ids = np.arange(n)
    for c in range(6):
        suffix[:, 5 - c] = (ids % 26) + ord('a')
        ids //= 26
    prefix = np.frombuffer(b"sharedpfx-", dtype=np.uint8)
    klen_content = len(prefix) + 6
    rec = 1 + klen_content + 4 + 8  # 1B vint + content + 4B BE vlen + 8B val
    view = np.zeros((n, rec), dtype=np.uint8)
    view[:, 0] = klen_content
    view[:, 1:1 + len(prefix)] = prefix
    view[:, 1 + len(prefix):1 + klen_content] = suffix
    view[:, 1 + klen_content:5 + klen_content] = np.frombuffer(
        np.int32(8).byteswap().tobytes(), dtype=np.uint8)
    view[:, 5 + klen_content:] = rng.integers(0, 256, size=(n, 8), dtype=np.uint8)
    data = view.reshape(-1)
    off = np.arange(0, rec * (n + 1), rec, dtype=np.uint64)
    klens = np.full(n, 1 + klen_content, dtype=np.uint32)
    dd, ofp, kp, _ = engine.upload_records(data.tobytes(), off, klens)
    conf = engine.make_conf(P, key_type=engine.KEY_TEXT,
                            comparator=engine.CMP_TEXT)
    s = engine.Sorter(conf)
    s.write_batch_device(dd, ofp, kp, None, n)
    s.flush()
    got, gidx = s.output()
    s.close()
    engine.free_device(dd, ofp, kp)
    want = o.spill(data, off, klens, P, key_type=o.KEY_TEXT,
                   comparator=o.CMP_TEXT)
    assert gidx == o.index_decode(want["index"], P)
    assert got == want["data"]


def test_varlen_tezbytes_multispill_documented_order(engine):
    """Variable-length TezBytes keys across spills: the reference's own merge
    output is emergent (proxy-then-serialized segment order is not
    comparator order — DESIGN.md par.3), so the engine pins the documented
    (partition, truncated proxy, serialized bytes) total order.  This test
    asserts that EXACT sequence, not just a multiset."""
    P = 4
    rng = random.Random(23)
    # include the documented divergence shape: share content prefixes across
    # different lengths ("aaaA" vs "aaaB" vs "aaa")
    pool = []
    seen = set()
    for base in (b"aaa", b"aaaA", b"aaaB", b"zz", b""):
        for i in range(120):
            k = base + bytes(rng.randrange(256) for _ in range(rng.randrange(0, 7)))
            if k in seen or len(k) == 0:
                continue
            seen.add(k)
            pool.append((o.serialize_bytes_writable(k),
                         o.serialize_bytes_writable(b"v%03d" % len(pool))))
    conf = engine.make_conf(P, key_type=engine.KEY_BYTES,
                            comparator=engine.CMP_TEZBYTES)
    s = engine.Sorter(conf)
    third = (len(pool) + 2) // 3
    for lo in range(0, len(pool), third):
        for k, v in pool[lo:lo + third]:
            s.write(k, v, -1)
        s.spill()
    assert s.num_spills() >= 2
    s.flush()
    got, gidx = s.output()
    s.close()
    assert _read_rows(got, gidx, P) == _expect_documented(pool, P)


def test_merge_rle_gate_same_key_events(engine):
    """The multi-spill auto-RLE gate counts SAME_KEY machine events (equal
    keys meeting across segments / RLE'd sources), NOT raw adjacent-equal
    pairs — TezMerger's emergent trace, restated by the oracle merge.  This
    shape sits in the divergence window: union adjacency 10.25% (a raw
    gate would enable RLE) but only 1.25% cross-segment events (the
    machine gate keeps it off), with every spill below its own gate."""
    P = 1
    val = o.serialize_bytes_writable

    def spill_keys(tag, ndup, nuniq):
        ks = []
        for i in range(ndup):
            k = o.serialize_bytes_writable(b"%c%03dDUP" % (tag, i))
            ks += [k, k]  # key appears twice within this spill
        for i in range(nuniq):
            ks.append(o.serialize_bytes_writable(b"%c%04dun" % (tag, i)))
        return ks

    shared = [o.serialize_bytes_writable(b"S%03dshr" % i) for i in range(50)]
    ka = spill_keys(ord('A'), 180, 2000 - 2 * 180 - 50) + shared
    kb = spill_keys(ord('B'), 180, 2000 - 2 * 180 - 50) + shared
    assert len(ka) == len(kb) == 2000
    conf = engine.make_conf(P, key_type=engine.KEY_BYTES,
                            comparator=engine.CMP_TEZBYTES)
    s = engine.Sorter(conf)
    spills = []
    for tag, keys in (("a", ka), ("b", kb)):
        pairs = [(k, val(b"%s%04d" % (tag.encode(), i)))
                 for i, k in enumerate(keys)]
        for k, v in pairs:
            s.write(k, v, -1)
        s.spill()
        d, f, kl = o.build_records(pairs)
        sp = o.spill(d, f, kl, P)
        assert sp["rle"] == 0  # each spill below its own gate
        spills.append(sp)
    s.flush()
    got, gidx = s.output()
    s.close()
    want = o.final_merge(spills, P)
    assert gidx == o.index_decode(want["index"], P)
    assert got == want["data"]


def test_add_sorted_segment_combiner(engine):
    """Reduce-side combine (MergeManager.java:916-921 runCombineProcessor):
    3 pre-sorted segments with shared keys, SUM_INT combiner at the merge —
    folded output equals the oracle's combined spill of the union."""
    import struct
    P = 2
    keys = [o.serialize_bytes_writable(b"ck%04d" % i) for i in range(300)]
    segs = []
    for sid in range(3):
        seg = [(k, struct.pack(">i", 1000 * sid + i))
               for i, k in enumerate(keys)]
        segs.append(_tezbytes_sorted(seg, P))
    conf = engine.make_conf(P, key_type=engine.KEY_BYTES,
                            comparator=engine.CMP_TEZBYTES, combiner=1,
                            min_spills_for_combine=3)
    s = engine.Sorter(conf)
    keep = []
    for seg in segs:
        d, f, kl = o.build_records(seg)
        dd, ofp, kp, _ = engine.upload_records(d.tobytes(), f, kl)
        keep.append((dd, ofp, kp))
        s.add_sorted_segment(dd, ofp, kp, None, len(seg))
    s.flush()
    got, gidx = s.output()
    s.close()
    union = [kv for seg in segs for kv in seg]
    d, f, kl = o.build_records(union)
    want = o.spill(d, f, kl, P, combiner=1)
    assert gidx == o.index_decode(want["index"], P)
    assert got == want["data"]
    for bufs in keep:
        engine.free_device(*bufs)
