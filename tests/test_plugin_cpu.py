"""CPU tests for the product-side IFile parser (tez_amd.ifile) against
hand-framed byte vectors (independent of both the engine and the oracle)."""
import zlib

from tez_amd import ifile


def frame(payload: bytes) -> bytes:
    return b"TIF\x00" + payload + zlib.crc32(payload).to_bytes(4, "big")


def test_vint_roundtrip():
    for v in (0, 1, -1, -2, -3, 127, -112, 128, 255, 256, -113, 2**31 - 1):
        enc = ifile.vint_write(v)
        dec, n = ifile.vint_read(enc, 0)
        assert dec == v and n == len(enc)


def test_plain_records():
    k1, v1 = b"\x00\x00\x00\x01A", b"\x00\x00\x00\x02xy"
    payload = bytes([5, 6]) + k1 + v1 + b"\xff\xff"
    recs = ifile.read_stream(frame(payload))
    assert recs == [(k1, v1, False)]


def test_rle_records():
    k = b"\x00\x00\x00\x01K"
    va, vb = b"\x00\x00\x00\x01a", b"\x00\x00\x00\x01b"
    payload = (bytes([5, 5]) + k + va
               + b"\xfe" + bytes([5]) + vb + b"\xfd" + b"\xff\xff")
    recs = ifile.read_stream(frame(payload))
    assert recs == [(k, va, False), (k, vb, True)]


def test_crc_rejected():
    payload = bytes([5, 5]) + b"\x00\x00\x00\x01K" + b"\x00\x00\x00\x01a" + b"\xff\xff"
    bad = bytearray(frame(payload))
    bad[6] ^= 1
    try:
        ifile.read_stream(bytes(bad))
        assert False, "should have raised"
    except ValueError:
        pass


def test_serde_helpers():
    assert ifile.deserialize_bytes_writable(ifile.serialize_bytes_writable(b"xy")) == b"xy"
    assert ifile.deserialize_text(ifile.serialize_text(b"word")) == b"word"
    assert ifile.deserialize_int_writable(ifile.serialize_int_writable(-7)) == -7
    assert ifile.serialize_int_writable(1) == b"\x00\x00\x00\x01"


def test_compressed_segment_roundtrip():
    """TIF\\1 framing: compress a plain segment, parse it
    back; matches the golden-fixture framing (CRC over compressed payload)."""
    import zlib
    k1, v1 = b"\x00\x00\x00\x01A", b"\x00\x00\x00\x02xy"
    payload = bytes([5, 6]) + k1 + v1 + b"\xff\xff"
    plain = frame(payload)
    comp = ifile.compress_segment(plain)
    assert comp[:4] == b"TIF\x01"
    assert zlib.crc32(comp[4:-4]) == int.from_bytes(comp[-4:], "big")
    assert ifile.read_stream(comp) == ifile.read_stream(plain)


def test_reference_golden_fixture_parses_with_product_reader():
    """The product reader must accept the reference's own compressed
    concatenated fixture segments."""
    import os
    FIX = ("/root/reference/tez-runtime-library/src/test/resources/"
           "TestIFile_concatenated_compressed.bin")
    if not os.path.exists(FIX):
        import pytest
        pytest.skip("reference tree absent")
    blob = open(FIX, "rb").read()
    comp = [723, 25396, 10926, 8203, 6665]
    pos = 0
    total = 0
    for c in comp:
        recs = ifile.read_stream(blob[pos:pos + c])
        total += len(recs)
        pos += c
    assert total > 100
