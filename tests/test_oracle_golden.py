"""Pin the oracle against the reference's own golden artifact:
tez-runtime-library/src/test/resources/TestIFile_concatenated_compressed.bin —
5 concatenated DefaultCodec(zlib) IFile streams with known raw/compressed
lengths (TestIFile.java:401-447 testConcatenatedZlibPadding).

This verifies, against reference-produced bytes: the TIF\\1 header flag, the
CRC32 trailer computed over the COMPRESSED payload (checksum stream sits below
the codec: IFile.java:352-368), the known length accounting, and the oracle's
record-framing parser (vints, RLE markers, EOF) on the decompressed payload
with Text keys + IntWritable values.

The fixture is read from /root/reference in THIS container only; a
repo-committed copy of the derived per-segment payloads is NOT needed because
this test is CPU-only and skipped where the reference tree is absent.
"""
import os
import zlib

import pytest

import oracle as o

FIXTURE = ("/root/reference/tez-runtime-library/src/test/resources/"
           "TestIFile_concatenated_compressed.bin")
RAWS = [2392, 102314, 42576, 31432, 25090]
COMPRESSED = [723, 25396, 10926, 8203, 6665]


@pytest.mark.skipif(not os.path.exists(FIXTURE), reason="reference tree absent")
def test_golden_concatenated_zlib():
    blob = open(FIXTURE, "rb").read()
    assert len(blob) == sum(COMPRESSED)
    pos = 0
    total_records = 0
    for raw_len, comp_len in zip(RAWS, COMPRESSED):
        seg = blob[pos: pos + comp_len]
        pos += comp_len
        # header: TIF, compressed flag = 1 (IFile.java:73-74,374-380)
        assert seg[:3] == b"TIF" and seg[3] == 1
        comp_payload = seg[4:-4]
        # CRC32 trailer over the compressed payload, big-endian
        # (IFileOutputStream.java:81-90 below the codec)
        assert int.from_bytes(seg[-4:], "big") == o.crc32(comp_payload)
        payload = zlib.decompress(comp_payload)
        # reduce-side accounting: the in-memory fetched segment
        # buffer is rawLength bytes, filled with rawLength-4 payload bytes
        assert len(payload) == raw_len - 4
        # parse with the oracle's reader (payload includes the EOF markers)
        recs = o.ifile_read(payload, with_header=False)
        assert len(recs) > 0
        for k, v, same in recs:
            # Text key: vint length + that many UTF-8 bytes
            klen, n = o.vint_decode(k)
            assert n + klen == len(k)
            # IntWritable value: exactly 4 bytes
            assert len(v) == 4
        total_records += len(recs)
    assert pos == len(blob)
    assert total_records > 100
