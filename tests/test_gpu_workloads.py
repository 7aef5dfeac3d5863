"""Workload-shaped parity on GPU for BASELINE configs C3 (Text/Zipf keys,
multi-spill merge) and C5 (TeraSort-shaped, range partitions), at
oracle-feasible sizes."""
import ctypes

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


def pull_generated(engine, d, off, kl, part, n):
    from tez_amd._engine import lib, _ck
    offs = np.zeros(n + 1, dtype=np.uint64)
    _ck(lib().tzs_memcpy_d2h(offs.ctypes.data, off, 8 * (n + 1)), "d2h")
    total = int(offs[-1])
    data = np.zeros(max(total, 1), dtype=np.uint8)
    if total:
        _ck(lib().tzs_memcpy_d2h(data.ctypes.data, d, total), "d2h")
    klen = np.zeros(n, dtype=np.uint32)
    _ck(lib().tzs_memcpy_d2h(klen.ctypes.data, kl, 4 * n), "d2h")
    parts = None
    if part:
        parts = np.zeros(n, dtype=np.int32)
        _ck(lib().tzs_memcpy_d2h(parts.ctypes.data, part, 4 * n), "d2h")
    return data, offs, klen, parts


def test_c5_terasort_shape_range_partitions(engine):
    """kind 2: 10B keys + 90B values, range partitioner (explicit partitions
    through the C-ABI, TotalOrderPartitioner-style)."""
    n, P = 30000, 16
    conf = engine.make_conf(P)
    d, off, kl, part = engine.generate(seed=0xC5, n=n, kind=2, klen=10, vlen=90,
                                       conf=conf)
    data, offs, klen, parts = pull_generated(engine, d, off, kl, part, n)
    # range partitioning: leading 2 key-content bytes
    for i in range(0, n, 977):
        b0, b1 = int(data[offs[i] + 4]), int(data[offs[i] + 5])
        assert parts[i] == ((b0 << 8 | b1) * P) >> 16
    s = engine.Sorter(conf)
    s.write_batch_device(d, off, kl, part, n)
    s.flush()
    got, gidx = s.output()
    s.close()
    engine.free_device(d, off, kl, part)
    want = o.spill(data, offs, klen, P, key_type=o.KEY_BYTES,
                   comparator=o.CMP_TEZBYTES, partitions=parts)
    assert gidx == o.index_decode(want["index"], P)
    assert got == want["data"]


def test_overflow_spill_count_coalesce(engine):
    """More spills than the RecTable's 32-slot merge table: the flush
    coalesces record sets (the reference reaches any count via multipass
    merges, getPassFactor/TezMerger.java:921-931).  Unique keys => byte-exact
    vs the oracle's merge of all 40 segments."""
    import random
    rng = random.Random(0x40)
    P, nspill, per = 8, 40, 120
    conf = engine.make_conf(P)
    s = engine.Sorter(conf)
    spills = []
    seen = set()
    for sp in range(nspill):
        pairs = []
        while len(pairs) < per:
            k = bytes(rng.randrange(256) for _ in range(12))
            if k in seen:
                continue
            seen.add(k)
            pairs.append((o.serialize_bytes_writable(k),
                          o.serialize_bytes_writable(b"v%02d.%03d" % (sp, len(pairs)))))
        for k, v in pairs:
            s.write(k, v, -1)
        s.spill()
        d, f, kl = o.build_records(pairs)
        spills.append(o.spill(d, f, kl, P))
    s.flush()
    got, gidx = s.output()
    ctr = s.counters()
    s.close()
    assert ctr["num_spills"] == nspill
    want = o.final_merge(spills, P)
    assert gidx == o.index_decode(want["index"], P)
    assert got == want["data"]


def test_c4_zipf_skewed_partitions(engine):
    """kind 3 (BASELINE configs[3] shape): uniform 10B keys whose PARTITION
    SIZES follow Zipf(1.0) over 199 partitions via the generator's
    inverse-CDF LUT; byte parity of the skewed sort vs the oracle."""
    import collections
    n, P = 30000, 199
    conf = engine.make_conf(P)
    d, off, kl, part = engine.generate(seed=0xC4, n=n, kind=3, klen=10, vlen=90,
                                       conf=conf)
    data, offs, klen, parts = pull_generated(engine, d, off, kl, part, n)
    cnt = collections.Counter(parts.tolist())
    # Zipf(1.0): partition 0 holds ~1/H(199) ~ 17%; heavy head, thin tail
    assert cnt[0] > n * 0.10
    assert cnt[0] > 5 * max(cnt.get(p, 0) for p in range(100, 199))
    s = engine.Sorter(conf)
    s.write_batch_device(d, off, kl, part, n)
    s.flush()
    got, gidx = s.output()
    s.close()
    engine.free_device(d, off, kl, part)
    want = o.spill(data, offs, klen, P, key_type=o.KEY_BYTES,
                   comparator=o.CMP_TEZBYTES, partitions=parts)
    assert gidx == o.index_decode(want["index"], P)
    assert got == want["data"]


def test_c3_text_zipf_multi_spill(engine):
    """kind 1: variable-length Text keys (Zipf word + unique suffix),
    4 spills merged at flush — the C3 shape at oracle scale."""
    n_per, P, nspill = 25000, 32, 4
    conf = engine.make_conf(P, key_type=engine.KEY_TEXT, comparator=engine.CMP_TEXT)
    s = engine.Sorter(conf)
    spills = []
    for k in range(nspill):
        d, off, kl, part = engine.generate(seed=0xC3 + k, n=n_per, kind=1,
                                           klen=0, vlen=64, conf=conf)
        data, offs, klen, _ = pull_generated(engine, d, off, kl, None, n_per)
        s.write_batch_device(d, off, kl, None, n_per)
        assert s.spill() == k
        engine.free_device(d, off, kl, part)
        spills.append(o.spill(data, offs, klen, P, key_type=o.KEY_TEXT,
                              comparator=o.CMP_TEXT))
        # per-spill segment view (pipelined shuffle):
        # spill files must byte-match the oracle spill too
        sdata, sidx = s.spill_output(k)
        assert sdata == spills[-1]["data"]
        assert sidx == o.index_decode(spills[-1]["index"], P)
    s.flush()
    got, gidx = s.output()
    ctr = s.counters()
    s.close()
    want = o.final_merge(spills, P, comparator=o.CMP_TEXT)
    assert gidx == o.index_decode(want["index"], P)
    assert got == want["data"]
    assert ctr["output_records"] == n_per * nspill
    # key-length variety sanity: the generator must produce short and long keys
    lens = set()
    for st, raw, cl in gidx:
        if cl:
            for kk, vv, _ in o.ifile_read(want["data"][st:st + cl], with_header=True):
                lens.add(len(kk))
    assert min(lens) >= 5 and max(lens) >= 15
