"""Device-side compressed IFile segments (TIF\\1) — SURVEY §8f row 2 /
VERDICT r1 #9.  The engine deflates each partition segment on device into
one zlib stream (32 KB chunks, fixed-Huffman LZ77 with stored fallback,
sync-flush stitched), CRC32 over the COMPRESSED payload — the reference's
DefaultCodec framing (IFile.java:352-368).  Compressed BYTES are
encoder-specific; the contract (validated here with an independent
inflater) is: valid zlib stream incl. adler32, decompressed payload
byte-identical to the uncompressed segment, CRC/partLength/rawLength
accounting per IFile.java:396-418."""
import random
import zlib

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


def _check_compressed(plain, pidx, comp, cidx, P):
    for p in range(P):
        st_u, raw_u, cl_u = pidx[p]
        st_c, raw_c, cl_c = cidx[p]
        assert raw_c == raw_u, f"p{p}: rawLength must keep uncompressed accounting"
        if cl_u == 0:
            assert cl_c == 0
            continue
        seg = comp[st_c:st_c + cl_c]
        assert seg[:4] == b"TIF\x01"
        body = seg[4:-4]  # zlib stream incl. header + adler32
        # CRC over the COMPRESSED payload (IFileOutputStream.finish)
        assert zlib.crc32(body) == int.from_bytes(seg[-4:], "big"), f"p{p} crc"
        # zlib.decompress verifies the adler32 trailer too
        payload = zlib.decompress(bytes(body))
        want = plain[st_u + 4:st_u + raw_u]  # body + tail, no CRC
        assert payload == bytes(want), f"p{p} payload mismatch"
        assert cl_c == 4 + len(body) + 4


def test_compressed_segments_roundtrip(engine):
    P = 8
    rng = random.Random(5)
    pairs = []
    for i in range(4000):
        k = bytes(rng.randrange(256) for _ in range(12))
        # compressible values: repeated text
        v = (b"value-%04d-" % (i % 50)) * 6
        pairs.append((o.serialize_bytes_writable(k), o.serialize_bytes_writable(v)))
    conf = engine.make_conf(P)
    s = engine.Sorter(conf)
    for k, v in pairs:
        s.write(k, v, -1)
    s.flush()
    plain, pidx = s.output()
    comp, cidx = s.output_compressed()
    s.close()
    _check_compressed(plain, pidx, comp, cidx, P)
    # compressible payload must actually compress
    tot_u = sum(r for _, r, _ in pidx)
    tot_c = sum(c for _, _, c in cidx)
    assert tot_c < 0.6 * tot_u, f"ratio {tot_c/tot_u:.2f} not compressive"
    # the product host reader consumes TIF\1 directly
    from tez_amd import ifile
    st, raw, cl = cidx[0]
    recs = ifile.read_stream(comp[st:st + cl])
    st_u, raw_u, cl_u = pidx[0]
    assert recs == ifile.read_stream(plain[st_u:st_u + cl_u])


def test_compressed_multichunk_and_incompressible(engine):
    """Segments far beyond one 32 KB deflate chunk (multi-chunk sync-flush
    stitching) and random (incompressible -> stored-block fallback) data."""
    P = 2
    n = 30000
    d, off, kl, part = engine.generate(seed=77, n=n, kind=0, klen=16, vlen=48,
                                       conf=engine.make_conf(P))
    s = engine.Sorter(engine.make_conf(P))
    s.write_batch_device(d, off, kl, None, n)
    s.flush()
    plain, pidx = s.output()
    comp, cidx = s.output_compressed()
    s.close()
    engine.free_device(d, off, kl, part)
    assert max(r for _, r, _ in pidx) > 3 * 32768  # multi-chunk segments
    _check_compressed(plain, pidx, comp, cidx, P)
    # random payload: stored fallback stays within ~0.1% overhead
    tot_u = sum(r for _, r, _ in pidx)
    tot_c = sum(c for _, _, c in cidx)
    assert tot_c < 1.01 * tot_u


def test_compressed_empty_and_rle(engine):
    P = 4
    key = o.serialize_bytes_writable(b"dup")
    s = engine.Sorter(engine.make_conf(P, rle=1))
    for i in range(40):
        s.write(key, o.serialize_bytes_writable(b"v%02d" % i), -1)
    s.flush()
    plain, pidx = s.output()
    comp, cidx = s.output_compressed()
    s.close()
    # all records share one key -> one partition present, others empty
    present = [p for p in range(P) if pidx[p][2] > 0]
    assert len(present) == 1
    _check_compressed(plain, pidx, comp, cidx, P)


def test_compressed_plugin_e2e(engine):
    """tez.runtime.compress=true through the plugin surface: the output
    emits TIF\\1 segments (device deflate), the grouped input consumes them
    unchanged (the reader inflates per the golden-fixture-pinned framing) —
    same groups as the uncompressed run."""
    from tez_amd.ordered_output import OrderedPartitionedKVOutput
    from tez_amd.ordered_input import OrderedGroupedKVInput

    P = 3
    base = {"tez.runtime.key.class": "org.apache.hadoop.io.Text",
            "tez.runtime.value.class": "org.apache.hadoop.io.IntWritable"}
    words = [f"word{i % 40:03d}" for i in range(5000)]

    def run(props):
        out = OrderedPartitionedKVOutput(P, props, unique_id="attempt_c0").start()
        w = out.get_writer()
        for word in words:
            w.write(word.encode(), 1)
        out.close()
        groups = {}
        for p in range(P):
            inp = OrderedGroupedKVInput(p, props)
            seg, raw = out.segment(p)
            if raw > 6:
                inp.add_segment(seg)
            inp.start()
            for key, vals in inp.get_reader():
                groups[key] = sum(int.from_bytes(v, "big") for v in vals)
        return out, groups

    out_c, got = run(dict(base, **{"tez.runtime.compress": "true"}))
    out_u, want = run(base)
    assert got == want
    assert sum(c for _, _, c in out_c._index) < sum(c for _, _, c in out_u._index)
    # rawLength (uncompressed accounting) must be identical
    assert [r for _, r, _ in out_c._index] == [r for _, r, _ in out_u._index]


def test_compressed_files_served_over_http(engine, tmp_path):
    """Compressed spill files through the /mapOutput protocol: the handler
    serves TIF\\1 partition ranges straight off file.out via the compressed
    index; the fetcher + reader inflate them back to the exact records
    (SURVEY §8f rows 2+3 composed)."""
    from tez_amd import shuffle_handler as sh
    from tez_amd import ifile
    P = 4
    rng = random.Random(9)
    pairs = []
    for i in range(3000):
        k = bytes(rng.randrange(256) for _ in range(10))
        pairs.append((o.serialize_bytes_writable(k),
                      o.serialize_bytes_writable((b"pay%03d" % (i % 40)) * 4)))
    s = engine.Sorter(engine.make_conf(P))
    for k, v in pairs:
        s.write(k, v, -1)
    s.flush()
    plain, pidx = s.output()
    s.write_files_compressed(str(tmp_path), "attempt_z0")
    s.close()
    srv = sh.ShuffleHandlerServer(str(tmp_path), port=0).start()
    try:
        for red in range(P):
            got = sh.fetch_map_outputs("127.0.0.1", srv.port, "job_1", "1",
                                       red, ["attempt_z0"])
            st_u, raw_u, cl_u = pidx[red]
            if raw_u <= 6:
                continue
            (mid, r2, rlen, seg), = got
            assert mid == "attempt_z0" and r2 == red
            assert rlen == raw_u  # rawLength keeps uncompressed accounting
            assert seg[:4] == b"TIF\x01"
            recs = ifile.read_stream(seg)
            assert recs == ifile.read_stream(plain[st_u:st_u + cl_u])
    finally:
        srv.stop()
