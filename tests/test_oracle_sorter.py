"""Semantic tests of the oracle's spill + final-merge restatement
(PipelinedSorter.java / TezMerger.java — see oracle/tzoracle.c citations).

The expected values here come from an INDEPENDENT in-test Python model of the
reference's sort order (prefix int, comparator), so the
oracle's C code is cross-checked, not self-checked.  A key cross-validation:
for unique keys, final_merge(k spills) must byte-equal spill(union) per
partition — two different algorithms, one answer.
"""
import random

import numpy as np
import pytest

import oracle as o


def py_prefix(part, nparts, content, has_proxy):
    pb = nparts.bit_length() + 1  # bitcount(P)+1, PipelinedSorter.java:165,317-324
    proxy = 0
    if has_proxy:
        b = list(content[:3]) + [0, 0, 0]
        proxy = (b[0] << 16) | (b[1] << 8) | b[2]
    return ((part << (32 - pb)) | (proxy >> pb)) & 0xFFFFFFFF


def py_cmp_key_tezbytes(a, b):
    # unsigned memcmp over the full serialized form, then length
    if a[: len(b)] != b[: len(a)][: len(a)]:
        pass
    m = min(len(a), len(b))
    if a[:m] != b[:m]:
        return -1 if a[:m] < b[:m] else 1
    return (len(a) > len(b)) - (len(a) < len(b))


def make_bytes_records(n, klen, vlen, seed, nparts):
    rng = random.Random(seed)
    pairs = []
    seen = set()
    while len(pairs) < n:
        k = bytes(rng.randrange(256) for _ in range(klen))
        if k in seen:
            continue
        seen.add(k)
        v = bytes(rng.randrange(256) for _ in range(vlen))
        pairs.append((o.serialize_bytes_writable(k), o.serialize_bytes_writable(v)))
    return pairs


def expected_order(ser_pairs, nparts, key_type, comparator):
    """Independent model: sort by (prefix, serialized-key comparator, index)."""
    items = []
    for i, (k, v) in enumerate(ser_pairs):
        if key_type == o.KEY_BYTES:
            content = k[4:]
        else:
            _, nn = o.vint_decode(k)
            content = k[nn:]
        part = (o.hash_bytes(content) & 0x7FFFFFFF) % nparts
        pref = py_prefix(part, nparts, content, comparator == o.CMP_TEZBYTES)
        items.append((pref, k, i, part))
    import functools

    def cmp(a, b):
        if a[0] != b[0]:
            return -1 if a[0] < b[0] else 1
        c = py_cmp_key_tezbytes(a[1], b[1]) if True else 0
        if c:
            return c
        return a[2] - b[2]
    items.sort(key=functools.cmp_to_key(cmp))
    return items


def test_spill_order_and_bytes_small():
    n, P = 500, 13
    pairs = make_bytes_records(n, klen=8, vlen=5, seed=7, nparts=P)
    data, off, klen = o.build_records(pairs)
    res = o.spill(data, off, klen, P, key_type=o.KEY_BYTES,
                  comparator=o.CMP_TEZBYTES, want_order=True)
    assert res["rle"] == 0
    model = expected_order(pairs, P, o.KEY_BYTES, o.CMP_TEZBYTES)
    assert list(res["order"]) == [i for _, _, i, _ in model]

    # reconstruct every partition stream and check contents + index arithmetic
    idx = o.index_decode(res["index"], P)
    dat = res["data"]
    by_part = {}
    for pref, k, i, part in model:
        by_part.setdefault(part, []).append(i)
    cursor = 0
    for p in range(P):
        start, raw, plen = idx[p]
        assert start == cursor
        recs_in_p = by_part.get(p, [])
        if not recs_in_p:
            assert raw == 0 and plen == 0
            continue
        seg = dat[start: start + plen]
        recs = o.ifile_read(seg, with_header=True)
        assert [k for k, v, s in recs] == [pairs[i][0] for i in recs_in_p]
        assert [v for k, v, s in recs] == [pairs[i][1] for i in recs_in_p]
        # a9: rawLength = 4 + sum(record framing) + 2
        body = sum(len(o.vint_encode(len(pairs[i][0]))) + len(o.vint_encode(len(pairs[i][1])))
                   + len(pairs[i][0]) + len(pairs[i][1]) for i in recs_in_p)
        assert raw == 4 + body + 2
        assert plen == raw + 4  # + CRC, uncompressed
        cursor += plen
    assert cursor == len(dat)


def test_spill_empty_partition_modes():
    # One record, 4 partitions: with send_empty, empty partitions are (0,0);
    # without, they are header+EOF+CRC streams (spill :586-601, rawLength 6,
    # partLength 10)
    pairs = [(o.serialize_bytes_writable(b"k"), o.serialize_bytes_writable(b"v"))]
    data, off, klen = o.build_records(pairs)
    res = o.spill(data, off, klen, 4, send_empty=True)
    idx = o.index_decode(res["index"], 4)
    nonempty = [t for t in idx if t[2] > 0]
    assert len(nonempty) == 1
    res2 = o.spill(data, off, klen, 4, send_empty=False)
    idx2 = o.index_decode(res2["index"], 4)
    for start, raw, plen in idx2:
        assert plen >= 10
        assert raw in (6, raw)
    empties = [t for t in idx2 if t[1] == 6]
    assert len(empties) == 3
    for start, raw, plen in empties:
        seg = res2["data"][start: start + plen]
        assert o.ifile_read(seg, with_header=True) == []


def test_spill_empty_input():
    data, off, klen = o.build_records([])
    res = o.spill(data, off, klen, 3, send_empty=True)
    assert res["data"] == b""
    assert o.index_decode(res["index"], 3) == [(0, 0, 0)] * 3


def test_spill_rle_auto_gate():
    # 100 records, only 3 distinct keys in 1 partition => adjacent-equal pairs
    # = 97 > 0.1*100 => rle on; stream must contain RLE markers
    key = o.serialize_bytes_writable(b"dupkey")
    pairs = [(key, o.serialize_bytes_writable(bytes([i]))) for i in range(100)]
    data, off, klen = o.build_records(pairs)
    res = o.spill(data, off, klen, 1)
    assert res["rle"] == 1
    idx = o.index_decode(res["index"], 1)
    recs = o.ifile_read(res["data"][: idx[0][2]], with_header=True)
    assert len(recs) == 100
    assert sum(1 for _, _, s in recs if s) == 99
    # values preserved in original order (stable tie-break)
    assert [v for _, v, _ in recs] == [p[1] for p in pairs]


def test_final_merge_equals_union_spill_unique_keys():
    """Cross-validation: merge(spills of chunks) == spill(union) for unique keys."""
    n, P = 1200, 7
    pairs = make_bytes_records(n, klen=10, vlen=6, seed=21, nparts=P)
    data, off, klen = o.build_records(pairs)
    union = o.spill(data, off, klen, P)

    spills = []
    for lo in range(0, n, 300):
        d, f, k = o.build_records(pairs[lo: lo + 300])
        spills.append(o.spill(d, f, k, P))
    merged = o.final_merge(spills, P)
    assert merged["data"] == union["data"]
    assert merged["index"] == union["index"]


def test_final_merge_multipass_factor():
    """k=10 segments with factor=3 forces the multi-pass path
    (getPassFactor, TezMerger.java:921-931); result must still equal the
    single-pass union for unique keys."""
    n, P = 800, 3
    pairs = make_bytes_records(n, klen=9, vlen=4, seed=33, nparts=P)
    union_d, union_o, union_k = o.build_records(pairs)
    union = o.spill(union_d, union_o, union_k, P)
    spills = []
    for lo in range(0, n, 80):
        d, f, k = o.build_records(pairs[lo: lo + 80])
        spills.append(o.spill(d, f, k, P))
    merged = o.final_merge(spills, P, factor=3)
    assert merged["data"] == union["data"]
    assert merged["index"] == union["index"]


def test_final_merge_rle_across_segments():
    """Same key in different spills: the merge must emit SAME_KEY/RLE runs
    (compareKeyWithNextTopKey, TezMerger.java:642-653)."""
    P = 1
    key = o.serialize_bytes_writable(b"sharedkey")
    s1_pairs = [(key, o.serialize_bytes_writable(b"s1-%d" % i)) for i in range(20)]
    s2_pairs = [(key, o.serialize_bytes_writable(b"s2-%d" % i)) for i in range(20)]
    spills = []
    for ps in (s1_pairs, s2_pairs):
        d, f, k = o.build_records(ps)
        spills.append(o.spill(d, f, k, P))
    merged = o.final_merge(spills, P)
    idx = o.index_decode(merged["index"], P)
    recs = o.ifile_read(merged["data"][: idx[0][2]], with_header=True)
    assert len(recs) == 40
    # exactly one full key in the stream; 39 SAME_KEY continuations
    assert sum(1 for _, _, s in recs if not s) == 1
    vals = set(v for _, v, _ in recs)
    assert len(vals) == 40


def test_text_comparator_order():
    """Text keys: content memcmp then shorter-first; no proxy in the prefix."""
    contents = [b"b", b"ab", b"a", b"aa", b"abc", b"", b"a\x00", b"z" * 30]
    pairs = [(o.serialize_text(c), o.serialize_bytes_writable(b"v")) for c in contents]
    data, off, klen = o.build_records(pairs)
    res = o.spill(data, off, klen, 1, key_type=o.KEY_TEXT,
                  comparator=o.CMP_TEXT, want_order=True)
    got = [contents[i] for i in res["order"]]
    # hadoop Text order == python bytes order except ties resolved by length
    # (python bytes compare IS memcmp-then-shorter-first)
    assert got == sorted(contents)


def test_spill_explicit_partitions():
    """Explicit per-record partitions (range/LUT partitioners — C4/C5 shapes)
    override HashPartitioner placement; order within a partition still follows
    the comparator, and the independent model agrees."""
    n, P = 800, 9
    pairs = make_bytes_records(n, klen=10, vlen=4, seed=5, nparts=P)
    import random as _r
    rng = _r.Random(5)
    parts = np.array([rng.randrange(P) for _ in range(n)], dtype=np.int32)
    data, off, klen = o.build_records(pairs)
    res = o.spill(data, off, klen, P, key_type=o.KEY_BYTES,
                  comparator=o.CMP_TEZBYTES, partitions=parts, want_order=True)
    # independent model with the EXPLICIT placement
    import functools
    items = []
    for i, (k, v) in enumerate(pairs):
        content = k[4:]
        pref = py_prefix(int(parts[i]), P, content, True)
        items.append((pref, k, i, int(parts[i])))

    def cmp(a, b):
        if a[0] != b[0]:
            return -1 if a[0] < b[0] else 1
        c = py_cmp_key_tezbytes(a[1], b[1])
        return c if c else a[2] - b[2]

    items.sort(key=functools.cmp_to_key(cmp))
    assert list(res["order"]) == [i for _, _, i, _ in items]
    idx = o.index_decode(res["index"], P)
    # every record landed in ITS partition
    cursor = 0
    import collections
    by_part = collections.Counter(parts)
    for p in range(P):
        start, raw, plen = idx[p]
        assert start == cursor
        seg = res["data"][start:start + plen]
        recs = o.ifile_read(seg, with_header=True) if plen else []
        assert len(recs) == by_part.get(p, 0)
        cursor += plen


def test_spill_mt_matches_single_thread():
    """The partition-parallel spill (the multi-core CPU baseline) must be
    byte-identical to the single-threaded restatement."""
    n, P = 3000, 13
    pairs = make_bytes_records(n, klen=12, vlen=9, seed=77, nparts=P)
    data, off, klen = o.build_records(pairs)
    want = o.spill(data, off, klen, P)
    for threads in (1, 3, 8):
        got = o.spill_mt(data, off, klen, P, threads)
        assert got["data"] == want["data"], threads
        assert got["index"] == want["index"], threads
