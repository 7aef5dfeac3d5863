"""Shuffle event payload codecs — wire-compatible with
tez-runtime-library/src/main/proto/ShufflePayloads.proto and the builders in
ShuffleUtils.generateEventOnSpill / generateDMEPayload
(ShuffleUtils.java:289-340,421-480).

Hand-encoded protobuf wire format (proto2 semantics: explicitly-set fields are
serialized even at default values, in field-number order — matching the
reference's builder output).  empty_partitions carries a java
BitSet.toByteArray (little-endian bit order) deflated with zlib level 9
(TezCommonUtils.newBestCompressionDeflater).
"""
import zlib


# ---- protobuf wire primitives ----

def _varint(v: int) -> bytes:
    out = bytearray()
    v &= (1 << 64) - 1
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _tag(field: int, wire: int) -> bytes:
    return _varint((field << 3) | wire)


def _len_delim(field: int, payload: bytes) -> bytes:
    return _tag(field, 2) + _varint(len(payload)) + payload


def _read_varint(b, pos):
    v = 0
    shift = 0
    while True:
        x = b[pos]
        pos += 1
        v |= (x & 0x7F) << shift
        if not x & 0x80:
            return v, pos
        shift += 7


def parse_message(b: bytes):
    """Generic wire parse -> {field: [values]}; values are ints or bytes."""
    out = {}
    pos = 0
    while pos < len(b):
        key, pos = _read_varint(b, pos)
        field, wire = key >> 3, key & 7
        if wire == 0:
            v, pos = _read_varint(b, pos)
        elif wire == 2:
            ln, pos = _read_varint(b, pos)
            v = b[pos: pos + ln]
            pos += ln
        elif wire == 5:
            v = int.from_bytes(b[pos: pos + 4], "little")
            pos += 4
        elif wire == 1:
            v = int.from_bytes(b[pos: pos + 8], "little")
            pos += 8
        else:
            raise ValueError(f"wire type {wire}")
        out.setdefault(field, []).append(v)
    return out


# ---- java BitSet codec (TezUtilsInternal.toByteArray / fromByteArray) ----

def bitset_to_bytes(bits) -> bytes:
    """java.util.BitSet.toByteArray: byte i holds bits 8i..8i+7, LSB first;
    trailing zero bytes dropped."""
    if not bits:
        return b""
    hi = max(bits)
    out = bytearray(hi // 8 + 1)
    for b in bits:
        out[b // 8] |= 1 << (b % 8)
    return bytes(out)


def bitset_from_bytes(b: bytes):
    return {i * 8 + j for i, byte in enumerate(b) for j in range(8) if byte >> j & 1}


# ---- DataMovementEventPayloadProto (ShufflePayloads.proto:23-34) ----

def build_dme_payload(index, host="", port=0, path_component="",
                      send_empty_partition_details=True, final_merge_enabled=True,
                      spill_id=0, last_event=True):
    """index: list of (start, raw_length, part_length) per partition
    (hasData iff raw_length > 6 — TezIndexRecord.java:52-56).
    Mirrors generateDMEPayload (ShuffleUtils.java:289-340)."""
    # fields serialized in field-number order (1,2,3,4,5,8,9) as the
    # reference builder does
    out = b""
    empty = {p for p, (_s, raw, _c) in enumerate(index) if raw <= 6}
    output_generated = len(empty) != len(index)
    if send_empty_partition_details and empty:
        out += _len_delim(1, zlib.compress(bitset_to_bytes(empty), 9))
    if not send_empty_partition_details or output_generated:
        out += _len_delim(2, host.encode())
        out += _tag(3, 0) + _varint(port)
        out += _len_delim(4, path_component.encode())
    out += _tag(5, 0) + _varint(0)  # run_duration, always explicitly 0
    if not final_merge_enabled:
        out += _tag(8, 0) + _varint(1 if last_event else 0)
        out += _tag(9, 0) + _varint(spill_id)
    return out


def parse_dme_payload(b: bytes, num_partitions=None):
    m = parse_message(b)
    out = {
        "host": m.get(2, [b""])[0].decode(),
        "port": m.get(3, [0])[0],
        "path_component": m.get(4, [b""])[0].decode(),
        "run_duration": m.get(5, [0])[0],
        "last_event": bool(m.get(8, [1])[0]),
        "spill_id": m.get(9, [None])[0],
        "empty_partitions": set(),
    }
    if 1 in m:
        out["empty_partitions"] = bitset_from_bytes(zlib.decompress(m[1][0]))
    return out


# ---- VertexManagerEventPayloadProto (ShufflePayloads.proto:52-57) ----

def build_vm_payload(output_size: int, num_record: int, partition_bytes=None):
    """generateVMEvent (ShuffleUtils.java:444-480) with detailed partition
    stats (DetailedPartitionStatsProto: repeated int32 size_in_mb, rounded
    UP to whole MB — ShuffleUtils.java:511-516)."""
    out = _tag(1, 0) + _varint(output_size)
    if partition_bytes is not None:
        inner = b"".join(_tag(1, 0) + _varint((sz + (1 << 20) - 1) >> 20)
                         for sz in partition_bytes)
        out += _len_delim(3, inner)
    out += _tag(4, 0) + _varint(num_record)
    return out


def parse_vm_payload(b: bytes):
    m = parse_message(b)
    out = {"output_size": m.get(1, [0])[0], "num_record": m.get(4, [0])[0]}
    if 3 in m:
        out["size_in_mb"] = parse_message(m[3][0]).get(1, [])
    return out


# ---- event objects (the AM routing contract, SURVEY §2 'Event payloads') ----

class CompositeDataMovementEvent:
    """CompositeDataMovementEvent.create(0, numPartitions, payload)
    (ShuffleUtils.java:439-441): one event covering source outputs
    [offset, offset+count)."""

    def __init__(self, source_index_start, count, payload: bytes):
        self.source_index_start = source_index_start
        self.count = count
        self.payload = payload


class VertexManagerEvent:
    def __init__(self, target_vertex: str, payload: bytes):
        self.target_vertex = target_vertex
        self.payload = payload


def events_on_flush(index, num_partitions, host, port, path_component,
                    output_size, num_record, partition_bytes=None,
                    send_empty_partition_details=True):
    """The List<Event> OrderedPartitionedKVOutput.close returns after a
    final-merge flush (PipelinedSorter keeps them in finalEvents;
    generateEventOnSpill ShuffleUtils.java:421-442): a VertexManagerEvent
    then a CompositeDataMovementEvent spanning all partitions."""
    vm = VertexManagerEvent("<dest>", build_vm_payload(output_size, num_record,
                                                       partition_bytes))
    dme = build_dme_payload(index, host, port, path_component,
                            send_empty_partition_details,
                            final_merge_enabled=True, last_event=True)
    return [vm, CompositeDataMovementEvent(0, num_partitions, dme)]
