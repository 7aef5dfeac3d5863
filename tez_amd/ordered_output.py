"""OrderedPartitionedKVOutput — the map-side plugin surface
(output/OrderedPartitionedKVOutput.java:92-219) over the HIP engine.

Lifecycle mirrors the reference: initialize (conf) -> start (sorter created,
:150-161) -> getWriter().write(k, v) (:168-181) -> close() -> flush + events
(:189-219).  Keys/values are raw CONTENT bytes; serialization to the
configured Writable form happens here (the reference serializes via the key
serializer at collect, PipelinedSorter.java:427-430).
"""
from . import _engine, ifile
from . import events as ev
from .conf import conf_from_tez_properties


class KeyValuesWriter:
    def __init__(self, out):
        self._out = out

    def write(self, key_content: bytes, value_content: bytes):
        o = self._out
        o._sorter.write(o._ser_key(key_content), o._ser_val(value_content), -1)
        o._records += 1
        # OUTPUT_BYTES counter is maintained by the engine (serialized k+v)


class OrderedPartitionedKVOutput:
    def __init__(self, num_partitions, props=None, unique_id="attempt_0_0000_0_00_000000_0_10003",
                 host="localhost", port=0):
        self.num_partitions = num_partitions
        self.props = dict(props or {})
        self.unique_id = unique_id
        self.host = host
        self.port = port
        self._sorter = None
        self._records = 0
        self._events = None
        self._final_merge = str(self.props.get(
            "tez.runtime.enable.final-merge.in.output", "true")).lower() \
            in ("1", "true", "yes")
        # tez.runtime.compress + codec (ExternalSorter.java:228): the engine
        # emits DefaultCodec (zlib) TIF\1 segments on device
        self._compress = str(self.props.get(
            "tez.runtime.compress", "false")).lower() in ("1", "true", "yes")
        codec = self.props.get(
            "tez.runtime.compress.codec",
            "org.apache.hadoop.io.compress.DefaultCodec")
        if self._compress and codec != \
                "org.apache.hadoop.io.compress.DefaultCodec":
            raise ValueError(f"unsupported compress.codec {codec} "
                             "(engine emits DefaultCodec/zlib)")
        if self._compress and not self._final_merge:
            raise ValueError("tez.runtime.compress with pipelined shuffle "
                             "(final merge off) is not supported: per-spill "
                             "segments are served uncompressed")
        key_cls = self.props.get("tez.runtime.key.class",
                                 "org.apache.hadoop.io.BytesWritable")
        if key_cls == "org.apache.hadoop.io.Text":
            self._ser_key = ifile.serialize_text
        else:
            self._ser_key = ifile.serialize_bytes_writable
        val_cls = self.props.get("tez.runtime.value.class",
                                 "org.apache.hadoop.io.BytesWritable")
        if val_cls == "org.apache.hadoop.io.Text":
            self._ser_val = ifile.serialize_text
        elif val_cls == "org.apache.hadoop.io.IntWritable":
            self._ser_val = ifile.serialize_int_writable
        else:
            self._ser_val = ifile.serialize_bytes_writable

    def start(self):
        conf = conf_from_tez_properties(self.props, self.num_partitions)
        self._sorter = _engine.Sorter(conf)
        return self

    def get_writer(self):
        return KeyValuesWriter(self)

    def spill(self):
        """Force a spill; in pipelined-shuffle mode
        (tez.runtime.enable.final-merge.in.output=false +
        tez.runtime.pipelined-shuffle.enabled=true) returns the per-spill
        CompositeDataMovementEvent (sendPipelinedShuffleEvents,
        PipelinedSorter.java:374-385)."""
        sid = self._sorter.spill()
        _, index = self._sorter.spill_output(sid)
        dme = ev.build_dme_payload(
            index, self.host, self.port, f"{self.unique_id}_{sid}",
            final_merge_enabled=False, spill_id=sid, last_event=False)
        return ev.CompositeDataMovementEvent(0, self.num_partitions, dme)

    def close(self):
        """flush + final merge; returns the List<Event> equivalent
        (VertexManagerEvent + CompositeDataMovementEvent).  With final merge
        disabled, one DME per spill is returned instead
        (PipelinedSorter.flush :709-726)."""
        self._sorter.flush()
        if not self._final_merge:
            ctr = self._sorter.counters()
            nspills = self._sorter.num_spills()
            events = []
            for sid in range(nspills):
                _, index = self._sorter.spill_output(sid)
                dme = ev.build_dme_payload(
                    index, self.host, self.port, f"{self.unique_id}_{sid}",
                    final_merge_enabled=False, spill_id=sid,
                    last_event=(sid == nspills - 1))
                events.append(ev.CompositeDataMovementEvent(
                    0, self.num_partitions, dme))
            events.insert(0, ev.VertexManagerEvent(
                "<dest>", ev.build_vm_payload(ctr["output_bytes"],
                                              ctr["output_records"])))
            self._events = events
            self._spill_data = [self._sorter.spill_output(s2)
                                for s2 in range(nspills)]
            self._sorter.close()
            self._sorter = None
            return events
        if self._compress:
            data, index = self._sorter.output_compressed()
        else:
            data, index = self._sorter.output()
        ctr = self._sorter.counters()
        self._events = ev.events_on_flush(
            index, self.num_partitions, self.host, self.port,
            self.unique_id, ctr["output_bytes"], ctr["output_records"],
            partition_bytes=[r for _s, r, _c in index])
        self._data = data
        self._index = index
        self._sorter.close()
        self._sorter = None
        return self._events

    # local-mode accessors (the DISK_DIRECT shortcut the input uses,
    # FetcherOrderedGrouped.java:193-205)
    def segment(self, partition):
        s, r, c = self._index[partition]
        return self._data[s: s + c], r

    def spill_segment(self, spill_id, partition):
        """Pipelined mode: partition slice of one spill's IFile stream."""
        data, index = self._spill_data[spill_id]
        s, r, c = index[partition]
        return data[s: s + c], r
