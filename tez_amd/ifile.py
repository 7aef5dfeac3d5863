"""Product-side IFile stream parser (KeyValuesReader support).

Restates the reader semantics of IFile.Reader
(IFile.java:877-1001: positionToNextRecord / RLE & V_END handling) and the
vint codec (hadoop WritableUtils).  Pure Python — used by the plugin input
to ingest fetched segments; the hot path (sort/merge/emit) never goes
through here.
"""
import zlib


def vint_read(b, pos):
    if pos >= len(b):
        raise ValueError("truncated IFile segment (vint past end)")
    first = b[pos]
    first_s = first - 256 if first >= 128 else first
    if first_s >= -112:
        return first_s, pos + 1
    ln = (-119 - first_s) if first_s < -120 else (-111 - first_s)
    v = 0
    for k in range(1, ln):
        v = (v << 8) | b[pos + k]
    if first_s < -120:
        v = ~v
    return v, pos + ln


def vint_write(v):
    if -112 <= v <= 127:
        return bytes([v & 0xFF])
    neg = v < 0
    if neg:
        v = ~v
    nbytes = 0
    t = v
    while t:
        t >>= 8
        nbytes += 1
    marker = (-120 - nbytes) if neg else (-112 - nbytes)
    return bytes([marker & 0xFF]) + v.to_bytes(nbytes, "big")


def read_stream(stream, with_header=True, verify_crc=True):
    """Parse one IFile stream; returns list of (key_ser, val_ser, same_key).
    Handles both TIF\0 (plain) and TIF\1 (DefaultCodec/zlib-compressed
    payload, CRC over the COMPRESSED bytes — IFile.java:352-368, pinned by
    the reference golden fixture)."""
    if with_header:
        if stream[:3] != b"TIF":
            raise ValueError("bad IFile magic")
        body = stream[4:-4]
        if verify_crc and zlib.crc32(body) != int.from_bytes(stream[-4:], "big"):
            raise ValueError("IFile CRC mismatch")
        if stream[3] == 1:
            body = zlib.decompress(bytes(body))
        elif stream[3] != 0:
            raise ValueError("unknown IFile compression flag")
    else:
        body = stream
    pos = 0
    out = []
    cur_key = b""
    prev_rle = False
    while True:
        if prev_rle:
            vlen, pos = vint_read(body, pos)
            if vlen == -3:  # V_END: fresh lengths follow
                klen, pos = vint_read(body, pos)
                vlen, pos = vint_read(body, pos)
            else:
                klen = -2
        else:
            klen, pos = vint_read(body, pos)
            vlen, pos = vint_read(body, pos)
        if klen == -1 and vlen == -1:
            break
        same = klen == -2
        if klen < -1 and not same:
            raise ValueError(f"corrupt IFile segment (klen marker {klen})")
        if not same:
            if klen < 0 or pos + klen > len(body):
                raise ValueError("truncated IFile segment (key past end)")
            cur_key = bytes(body[pos: pos + klen])
            pos += klen
        if vlen < 0 or pos + vlen > len(body):
            raise ValueError("truncated IFile segment (value past end)")
        val = bytes(body[pos: pos + vlen])
        pos += vlen
        out.append((cur_key, val, same))
        prev_rle = same
    return out


def serialize_bytes_writable(content: bytes) -> bytes:
    return len(content).to_bytes(4, "big") + content


def deserialize_bytes_writable(ser: bytes) -> bytes:
    return ser[4:]


def serialize_text(content: bytes) -> bytes:
    return vint_write(len(content)) + content


def deserialize_text(ser: bytes) -> bytes:
    n, pos = vint_read(ser, 0)
    return bytes(ser[pos:pos + n])


def serialize_int_writable(v: int) -> bytes:
    return int(v).to_bytes(4, "big", signed=True)


def deserialize_int_writable(ser: bytes) -> int:
    return int.from_bytes(ser[:4], "big", signed=True)


def compress_segment(plain_stream: bytes) -> bytes:
    """Re-frame a plain TIF\0 segment as TIF\1 with DefaultCodec(zlib)
    payload (checksum below the codec,
    IFile.java:352-368).  rawLength of the segment is unchanged; the new
    partLength is the returned length."""
    assert plain_stream[:4] == b"TIF\x00"
    payload = zlib.compress(plain_stream[4:-4], 9)
    return b"TIF\x01" + payload + zlib.crc32(payload).to_bytes(4, "big")
