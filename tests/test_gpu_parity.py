"""GPU parity tests: the HIP engine's IFile output (data + index) must be
byte-identical to the CPU oracle on the same inputs (unique-key
inputs bit-exact, duplicate-key inputs deterministic-tie parity between the
two implementations).  All tests marked gpu."""
import os
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


def _mk_fixed(n, klen, vlen, seed, unique=True):
    rng = random.Random(seed)
    seen = set()
    pairs = []
    while len(pairs) < n:
        k = bytes(rng.randrange(256) for _ in range(klen))
        if unique:
            if k in seen:
                continue
            seen.add(k)
        v = bytes(rng.randrange(256) for _ in range(vlen))
        pairs.append((o.serialize_bytes_writable(k), o.serialize_bytes_writable(v)))
    return pairs


def _run_engine_host_path(engine, pairs, P, key_type, comparator, **conf_kw):
    conf = engine.make_conf(P, key_type=key_type, comparator=comparator, **conf_kw)
    s = engine.Sorter(conf)
    for k, v in pairs:
        s.write(k, v, -1)
    s.flush()
    data, idx = s.output()
    ctr = s.counters()
    s.close()
    return data, idx, ctr


def _oracle_single_spill(pairs, P, key_type, comparator, **kw):
    data, off, klen = o.build_records(pairs)
    res = o.spill(data, off, klen, P, key_type=key_type, comparator=comparator, **kw)
    return res


def test_parity_fixed_bytes_keys(engine):
    pairs = _mk_fixed(5000, 16, 64, seed=1)
    got, gidx, ctr = _run_engine_host_path(engine, pairs, 64, engine.KEY_BYTES,
                                           engine.CMP_TEZBYTES)
    want = _oracle_single_spill(pairs, 64, o.KEY_BYTES, o.CMP_TEZBYTES)
    assert gidx == o.index_decode(want["index"], 64)
    assert got == want["data"]
    assert ctr["output_records"] == 5000
    assert ctr["output_bytes"] == sum(len(k) + len(v) for k, v in pairs)


def test_parity_one_partition(engine):
    pairs = _mk_fixed(1000, 8, 5, seed=2)
    got, gidx, _ = _run_engine_host_path(engine, pairs, 1, engine.KEY_BYTES,
                                         engine.CMP_TEZBYTES)
    want = _oracle_single_spill(pairs, 1, o.KEY_BYTES, o.CMP_TEZBYTES)
    assert got == want["data"]
    assert gidx == o.index_decode(want["index"], 1)


def test_parity_empty_partitions_and_modes(engine):
    pairs = _mk_fixed(3, 16, 8, seed=3)
    for send_empty in (1, 0):
        got, gidx, _ = _run_engine_host_path(engine, pairs, 16, engine.KEY_BYTES,
                                             engine.CMP_TEZBYTES,
                                             send_empty_partition_details=send_empty)
        want = _oracle_single_spill(pairs, 16, o.KEY_BYTES, o.CMP_TEZBYTES,
                                    send_empty=bool(send_empty))
        assert got == want["data"], f"send_empty={send_empty}"
        assert gidx == o.index_decode(want["index"], 16)


def test_parity_single_record_and_empty_values(engine):
    pairs = [(o.serialize_bytes_writable(b"onlykey\x00\xff"),
              o.serialize_bytes_writable(b""))]
    got, gidx, _ = _run_engine_host_path(engine, pairs, 4, engine.KEY_BYTES,
                                         engine.CMP_TEZBYTES)
    want = _oracle_single_spill(pairs, 4, o.KEY_BYTES, o.CMP_TEZBYTES)
    assert got == want["data"]


def test_parity_empty_input(engine):
    got, gidx, _ = _run_engine_host_path(engine, [], 5, engine.KEY_BYTES,
                                         engine.CMP_TEZBYTES)
    assert gidx == [(0, 0, 0)] * 5
    assert got == b""


def test_parity_duplicates_rle(engine):
    """Duplicate-heavy input: auto-RLE fires; engine and oracle agree byte-
    for-byte (both stable by input order; engine rule == oracle rule)."""
    rng = random.Random(9)
    keys = [bytes(rng.randrange(256) for _ in range(12)) for _ in range(20)]
    pairs = []
    for i in range(2000):
        k = keys[rng.randrange(len(keys))]
        pairs.append((o.serialize_bytes_writable(k),
                      o.serialize_bytes_writable(b"v%06d" % i)))
    got, gidx, _ = _run_engine_host_path(engine, pairs, 8, engine.KEY_BYTES,
                                         engine.CMP_TEZBYTES)
    want = _oracle_single_spill(pairs, 8, o.KEY_BYTES, o.CMP_TEZBYTES)
    assert want["rle"] == 1
    assert got == want["data"]
    assert gidx == o.index_decode(want["index"], 8)


def test_parity_text_keys(engine):
    """Text keys (vint+utf8, Text.Comparator: content memcmp then shorter
    first), variable lengths 1..40 incl. long shared prefixes to force
    refinement levels."""
    rng = random.Random(11)
    words = [b"alpha", b"alphabet", b"alphabetical", b"a", b"", b"zebra",
             b"zebr", b"\x00\x00", b"\x00"]
    pairs = []
    for i in range(3000):
        w = words[rng.randrange(len(words))]
        suffix = bytes(rng.randrange(97, 123) for _ in range(rng.randrange(0, 30)))
        content = (w + suffix)[:40]
        pairs.append((o.serialize_text(content),
                      o.serialize_bytes_writable(b"v%d" % i)))
    got, gidx, _ = _run_engine_host_path(engine, pairs, 16, engine.KEY_TEXT,
                                         engine.CMP_TEXT)
    want = _oracle_single_spill(pairs, 16, o.KEY_TEXT, o.CMP_TEXT)
    assert got == want["data"]
    assert gidx == o.index_decode(want["index"], 16)


def test_parity_multi_spill_final_merge(engine):
    """3 spills merged at flush must equal the oracle's TezMerger restatement
    (cross-validated against spill-of-union in the CPU suite)."""
    pairs = _mk_fixed(3000, 16, 24, seed=17)
    conf = engine.make_conf(32, key_type=engine.KEY_BYTES,
                            comparator=engine.CMP_TEZBYTES)
    s = engine.Sorter(conf)
    spills = []
    for lo in range(0, 3000, 1000):
        for k, v in pairs[lo: lo + 1000]:
            s.write(k, v, -1)
        s.spill()
        d, f, kl = o.build_records(pairs[lo: lo + 1000])
        spills.append(o.spill(d, f, kl, 32))
    s.flush()
    got, gidx = s.output()
    s.close()
    want = o.final_merge(spills, 32)
    assert gidx == o.index_decode(want["index"], 32)
    assert got == want["data"]


def test_parity_duplicates_across_spills(engine):
    """Same keys in different spills: SAME_KEY/RLE provenance rule
     must match the oracle's MergeQueue state machine."""
    keyA = o.serialize_bytes_writable(b"shared-key-A")
    keyB = o.serialize_bytes_writable(b"shared-key-B")
    s1 = [(keyA, o.serialize_bytes_writable(b"s1a%d" % i)) for i in range(5)]
    s2 = [(keyA, o.serialize_bytes_writable(b"s2a%d" % i)) for i in range(5)] + \
         [(keyB, o.serialize_bytes_writable(b"s2b%d" % i)) for i in range(5)]
    conf = engine.make_conf(2, key_type=engine.KEY_BYTES,
                            comparator=engine.CMP_TEZBYTES)
    s = engine.Sorter(conf)
    spills = []
    for batch in (s1, s2):
        for k, v in batch:
            s.write(k, v, -1)
        s.spill()
        d, f, kl = o.build_records(batch)
        spills.append(o.spill(d, f, kl, 2))
    s.flush()
    got, gidx = s.output()
    s.close()
    want = o.final_merge(spills, 2)
    assert gidx == o.index_decode(want["index"], 2)
    assert got == want["data"]


def test_device_generator_matches_oracle_pipeline(engine):
    """tzs_generate → engine sort must equal oracle.spill on the generator's
    own bytes (pulled to host)."""
    import ctypes
    from tez_amd._engine import lib, _ck
    n, P, klen, vlen = 20000, 13, 16, 32
    conf = engine.make_conf(P)
    d, off, kl, part = engine.generate(seed=42, n=n, kind=0, klen=klen, vlen=vlen,
                                       conf=conf)
    s = engine.Sorter(conf)
    s.write_batch_device(d, off, kl, None, n)
    s.flush()
    got, gidx = s.output()
    s.close()
    rec = 4 + klen + 4 + vlen
    host = bytearray(rec * n)
    ba = (ctypes.c_char * len(host)).from_buffer(host)
    _ck(lib().tzs_memcpy_d2h(ctypes.addressof(ba), d, len(host)), "d2h")
    engine.free_device(d, off, kl, part)
    data = np.frombuffer(bytes(host), dtype=np.uint8).copy()
    offs = np.arange(0, rec * (n + 1), rec, dtype=np.uint64)
    klens = np.full(n, 4 + klen, dtype=np.uint32)
    want = o.spill(data, offs, klens, P)
    assert gidx == o.index_decode(want["index"], P)
    assert got == want["data"]


def test_parity_adopted_batch(engine):
    """Zero-copy absorb (tzs_sorter_write_batch_device_adopt): the sorter
    takes ownership of tzs-allocated device buffers; output must byte-equal
    the copying path / oracle.  Also checks the guard: adopting into a
    non-empty sorter must fail."""
    import numpy as np
    pairs = _mk_fixed(4000, 16, 64, seed=59)
    data, offs, klens = o.build_records(pairs)
    d, doff, dkl, _ = engine.upload_records(bytes(data), offs, klens)
    conf = engine.make_conf(64)
    s = engine.Sorter(conf)
    s.write_batch_device_adopt(d, doff, dkl, None, len(pairs))
    s.flush()
    got, gidx = s.output()
    s.close()
    want = _oracle_single_spill(pairs, 64, o.KEY_BYTES, o.CMP_TEZBYTES)
    assert gidx == o.index_decode(want["index"], 64)
    assert got == want["data"]

    # guard: second adopt into a non-empty sorter fails loudly
    d2, doff2, dkl2, _ = engine.upload_records(bytes(data), offs, klens)
    s2 = engine.Sorter(engine.make_conf(8))
    s2.write_batch_device(d2, doff2, dkl2, None, len(pairs))
    try:
        s2.write_batch_device_adopt(d2, doff2, dkl2, None, len(pairs))
        assert False, "adopt into non-empty sorter should fail"
    except RuntimeError:
        pass
    s2.close()
    engine.free_device(d2, doff2, dkl2)

    # guard: foreign (non-registry) pointers are rejected
    s3 = engine.Sorter(engine.make_conf(8))
    try:
        s3.write_batch_device_adopt(12345678, 2345678, 345678, None, 10)
        assert False, "foreign pointers should be rejected"
    except RuntimeError:
        pass
    s3.close()


def test_cross_thread_write_then_flush(engine):
    """the producer thread may differ from the flush thread
    (ExternalSorter threading).  Writes happen on a worker thread, flush on
    the main thread; output must equal the single-thread path."""
    import threading
    pairs = _mk_fixed(2000, 16, 32, seed=71)
    conf = engine.make_conf(8)
    s = engine.Sorter(conf)
    err = []

    def producer():
        try:
            for k, v in pairs:
                s.write(k, v, -1)
        except Exception as e:   # pragma: no cover
            err.append(e)

    t = threading.Thread(target=producer)
    t.start()
    t.join()
    assert not err
    s.flush()
    got, gidx = s.output()
    s.close()
    want = _oracle_single_spill(pairs, 8, o.KEY_BYTES, o.CMP_TEZBYTES)
    assert gidx == o.index_decode(want["index"], 8)
    assert got == want["data"]


def test_write_files_reference_layout(engine, tmp_path):
    pairs = _mk_fixed(100, 16, 16, seed=23)
    conf = engine.make_conf(4)
    s = engine.Sorter(conf)
    for k, v in pairs:
        s.write(k, v, -1)
    s.flush()
    uid = "attempt_1_0001_1_00_000000_0_10003"
    s.write_files(str(tmp_path), uid)
    data, idx = s.output()
    s.close()
    fo = tmp_path / "output" / uid / "file.out"
    fi = tmp_path / "output" / uid / "file.out.index"
    assert fo.read_bytes() == data
    parsed = o.index_decode(fi.read_bytes(), 4)
    assert parsed == idx


def test_parity_variable_length_bytes_keys(engine):
    """Variable-length BytesWritable keys: the reference order is 3-byte
    content proxy FIRST, then the serialized compare (4B length prefix =>
    length-then-content) — PipelinedSorter.java:451-457 +
    TezBytesComparator.  Includes the divergence case where content order
    and serialized order disagree ('aaaAx' vs 'aaaB')."""
    rng = random.Random(41)
    contents = [b"aaaAx", b"aaaB", b"\x01", b"\x00\x00", b"", b"a", b"a\x00",
                b"aaa", b"aaaZ" * 3, b"zz"]
    pairs = []
    for i in range(4000):
        c = contents[rng.randrange(len(contents))] + \
            (bytes([rng.randrange(256)]) * rng.randrange(0, 3))
        pairs.append((o.serialize_bytes_writable(c),
                      o.serialize_bytes_writable(b"v%05d" % i)))
    # plus random unique variable-length keys
    for i in range(4000):
        c = bytes(rng.randrange(256) for _ in range(rng.randrange(0, 24)))
        pairs.append((o.serialize_bytes_writable(c),
                      o.serialize_bytes_writable(b"w%05d" % i)))
    got, gidx, _ = _run_engine_host_path(engine, pairs, 8, engine.KEY_BYTES,
                                         engine.CMP_TEZBYTES)
    want = _oracle_single_spill(pairs, 8, o.KEY_BYTES, o.CMP_TEZBYTES)
    assert gidx == o.index_decode(want["index"], 8)
    assert got == want["data"]
