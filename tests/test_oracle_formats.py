"""Pin the oracle's format restatements against hand-computed byte vectors
(independent of the oracle's own code) and against round-trips.

References: IFile.java, IFileOutputStream.java:81-90, TezSpillRecord.java:112-147,
hadoop WritableUtils vint (SURVEY.md §8c third-party boundary)."""
import ctypes
import zlib

import numpy as np
import pytest

import oracle as o


def test_vint_hand_vectors():
    # WritableUtils.writeVLong restated by hand:
    # [-112,127] one byte, two's complement
    assert o.vint_encode(0) == b"\x00"
    assert o.vint_encode(1) == b"\x01"
    assert o.vint_encode(127) == b"\x7f"
    assert o.vint_encode(-1) == b"\xff"          # EOF_MARKER
    assert o.vint_encode(-2) == b"\xfe"          # RLE_MARKER
    assert o.vint_encode(-3) == b"\xfd"          # V_END_MARKER
    assert o.vint_encode(-112) == b"\x90"
    # 128: marker -113 (0x8f), then 0x80
    assert o.vint_encode(128) == b"\x8f\x80"
    # 255 -> 0x8f 0xff ; 256 -> 0x8e 0x01 0x00
    assert o.vint_encode(255) == b"\x8f\xff"
    assert o.vint_encode(256) == b"\x8e\x01\x00"
    # negative beyond -112: -113 -> ~(-113)=112 -> marker -121 (0x87), 0x70
    assert o.vint_encode(-113) == b"\x87\x70"
    # 90 (C2's record body size) is a single byte
    assert o.vint_encode(90) == bytes([90])


def test_crc32_is_iso_hdlc():
    # zlib's crc32 IS the ISO-HDLC CRC that hadoop DataChecksum.CRC32 and
    # PureJavaCrc32 implement; "123456789" -> 0xCBF43926 is the standard KAT.
    assert o.crc32(b"123456789") == 0xCBF43926
    assert o.crc32(b"") == 0
    data = bytes(range(256)) * 17
    assert o.crc32(data) == zlib.crc32(data)


def test_java_hash_bytes():
    # h = 1; h = 31*h + signed(byte)   (hadoop WritableComparator.hashBytes)
    assert o.hash_bytes(b"") == 1
    assert o.hash_bytes(b"\x00") == 31
    assert o.hash_bytes(b"a") == 31 + 97
    assert o.hash_bytes(b"abc") == 31 * (31 * (31 + 97) + 98) + 99
    # signed bytes: 0xFF == -1
    assert o.hash_bytes(b"\xff") == 31 - 1
    # partition clamps with & Integer.MAX_VALUE (HashPartitioner.java:32-35)
    assert 0 <= o.partition_of(b"\xff\xfe\xfd", 7) < 7


def test_ifile_writer_hand_vector_plain():
    """Two unique records, no RLE: stream must be exactly
    TIF\\0 | {vint k, vint v, key, val} x2 | -1 -1 | CRC32(payload)."""
    w = o._lib.tzo_writer_new(0)
    k1, v1 = b"\x00\x00\x00\x01A", b"\x00\x00\x00\x02xy"
    k2, v2 = b"\x00\x00\x00\x01B", b"\x00\x00\x00\x01z"
    for k, v in [(k1, v1), (k2, v2)]:
        ka = np.frombuffer(k, dtype=np.uint8).copy()
        va = np.frombuffer(v, dtype=np.uint8).copy()
        o._lib.tzo_writer_append(w, o._u8p(ka), len(k), o._u8p(va), len(v))
    out, ln, raw, part = (ctypes.c_void_p(), ctypes.c_int64(), ctypes.c_int64(), ctypes.c_int64())
    o._lib.tzo_writer_close(w, ctypes.byref(out), ctypes.byref(ln), ctypes.byref(raw), ctypes.byref(part))
    stream = ctypes.string_at(out.value, ln.value)
    o._lib.tzo_free(out.value)
    o._lib.tzo_writer_free(w)

    payload = (bytes([5, 6]) + k1 + v1 + bytes([5, 5]) + k2 + v2 + b"\xff\xff")
    expect = b"TIF\x00" + payload + zlib.crc32(payload).to_bytes(4, "big")
    assert stream == expect
    # a9 accounting: rawLength = header + records + EOF (no CRC);
    # partLength = header + payload + CRC
    assert raw.value == 4 + len(payload)
    assert part.value == len(stream)


def test_ifile_writer_hand_vector_rle():
    """RLE framing: {k,v1},{RLE,v2},{V_END} then EOF (IFile.java:590-615)."""
    w = o._lib.tzo_writer_new(1)
    k = b"\x00\x00\x00\x01K"
    v1, v2 = b"\x00\x00\x00\x01a", b"\x00\x00\x00\x01b"
    for v in [v1, v2]:
        ka = np.frombuffer(k, dtype=np.uint8).copy()
        va = np.frombuffer(v, dtype=np.uint8).copy()
        o._lib.tzo_writer_append(w, o._u8p(ka), len(k), o._u8p(va), len(v))
    out, ln, raw, part = (ctypes.c_void_p(), ctypes.c_int64(), ctypes.c_int64(), ctypes.c_int64())
    o._lib.tzo_writer_close(w, ctypes.byref(out), ctypes.byref(ln), ctypes.byref(raw), ctypes.byref(part))
    stream = ctypes.string_at(out.value, ln.value)
    o._lib.tzo_free(out.value)
    o._lib.tzo_writer_free(w)

    payload = (bytes([5, 5]) + k + v1          # first KV pair
               + b"\xfe" + bytes([5]) + v2     # RLE marker + vlen + value
               + b"\xfd"                       # V_END closes the run
               + b"\xff\xff")                  # EOF
    expect = b"TIF\x00" + payload + zlib.crc32(payload).to_bytes(4, "big")
    assert stream == expect


def test_ifile_reader_roundtrip_including_rle():
    pairs = [(b"k1", b"va"), (b"k2", b"v1"), (b"k2", b"v2"), (b"k2", b"v3"),
             (b"k3", b""), (b"", b"emptykey")]
    ser = [(o.serialize_bytes_writable(k), o.serialize_bytes_writable(v)) for k, v in pairs]
    w = o._lib.tzo_writer_new(1)
    for k, v in ser:
        ka = np.frombuffer(k, dtype=np.uint8).copy()
        va = np.frombuffer(v, dtype=np.uint8).copy()
        o._lib.tzo_writer_append(w, o._u8p(ka), len(k), o._u8p(va), len(v))
    out, ln, raw, part = (ctypes.c_void_p(), ctypes.c_int64(), ctypes.c_int64(), ctypes.c_int64())
    o._lib.tzo_writer_close(w, ctypes.byref(out), ctypes.byref(ln), ctypes.byref(raw), ctypes.byref(part))
    stream = ctypes.string_at(out.value, ln.value)
    o._lib.tzo_free(out.value)
    o._lib.tzo_writer_free(w)

    recs = o.ifile_read(stream, with_header=True)
    assert [(k, v) for k, v, _ in recs] == ser
    # records 3,4 (0-based 2,3) were SAME_KEY-encoded
    assert [s for _, _, s in recs] == [False, False, True, True, False, False]


def test_ifile_reader_rejects_bad_crc():
    w = o._lib.tzo_writer_new(0)
    ka = np.frombuffer(b"\x00\x00\x00\x01A", dtype=np.uint8).copy()
    o._lib.tzo_writer_append(w, o._u8p(ka), 5, o._u8p(ka), 5)
    out, ln, raw, part = (ctypes.c_void_p(), ctypes.c_int64(), ctypes.c_int64(), ctypes.c_int64())
    o._lib.tzo_writer_close(w, ctypes.byref(out), ctypes.byref(ln), ctypes.byref(raw), ctypes.byref(part))
    stream = bytearray(ctypes.string_at(out.value, ln.value))
    o._lib.tzo_free(out.value)
    o._lib.tzo_writer_free(w)
    stream[7] ^= 0x40
    with pytest.raises(ValueError):
        o.ifile_read(bytes(stream), with_header=True)


def test_index_encode_hand_vector():
    triples = np.array([0, 6, 10, 10, 100, 110], dtype=np.int64)
    out = np.zeros(24 * 2 + 8, dtype=np.uint8)
    n = o._lib.tzo_index_encode(triples.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)), 2, o._u8p(out))
    assert n == 56
    body = b"".join(int(x).to_bytes(8, "big") for x in triples)
    assert out[:48].tobytes() == body
    assert out[48:].tobytes() == zlib.crc32(body).to_bytes(8, "big")
    parsed = o.index_decode(out.tobytes(), 2)
    assert parsed == [(0, 6, 10), (10, 100, 110)]


def test_shuffle_header_roundtrip():
    b = o.shuffle_header_encode("attempt_123_0001_1_00_000000_0_10003", 72345, 98765, 17)
    mid, clen, rlen, part, n = o.shuffle_header_decode(b)
    assert (mid, clen, rlen, part) == ("attempt_123_0001_1_00_000000_0_10003", 72345, 98765, 17)
    assert n == len(b)
    # framing: vint(idlen) + id + vlong clen + vlong rlen + vint partition
    assert b[0] == 36 and b[1:37] == b"attempt_123_0001_1_00_000000_0_10003"
