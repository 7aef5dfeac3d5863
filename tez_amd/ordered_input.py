"""OrderedGroupedKVInput — the reduce-side plugin surface
(input/OrderedGroupedKVInput.java:101-318) over the HIP engine.

Segments for THIS input's partition are added from producer outputs (the
local DISK_DIRECT path, FetcherOrderedGrouped.java:193-205, or the xGMI
exchange); start() runs the merge (Shuffle.run -> MergeManager.finalMerge
equivalents); get_reader() yields (key_content, [value_content...]) groups
with ValuesIterator semantics (ValuesIterator.java:177-199: same group iff
the merge reported SAME_KEY or the comparator says equal).
"""
from . import _engine, ifile


class OrderedGroupedKVInput:
    def __init__(self, partition, props=None):
        self.partition = partition
        self.props = dict(props or {})
        self._segments = []
        self._merged = None
        key_cls = self.props.get("tez.runtime.key.class",
                                 "org.apache.hadoop.io.BytesWritable")
        self._text_keys = key_cls == "org.apache.hadoop.io.Text"

    def add_segment(self, ifile_bytes, raw_length=None):
        """A fetched segment: IFile stream bytes of this partition from one
        producer (ShuffleHeader-framed on the wire; the frame is handled by
        the exchange layer)."""
        if len(ifile_bytes) == 0:
            return
        self._segments.append(ifile_bytes)

    def start(self):
        """Merge the segments through the engine (columnar ingestion +
        stable re-sort — DESIGN.md §4).  One partition in, one partition
        out."""
        pairs = []
        for seg in self._segments:
            for k, v, _same in ifile.read_stream(seg):
                pairs.append((k, v))
        conf = _engine.make_conf(
            1,  # single-partition merge; placement already decided map-side
            key_type=_engine.KEY_TEXT if self._text_keys else _engine.KEY_BYTES,
            comparator=_engine.CMP_TEXT if self._text_keys else _engine.CMP_TEZBYTES)
        s = _engine.Sorter(conf)
        for k, v in pairs:
            s.write(k, v, -1)  # hash%1 == 0; avoids the explicit-partition path
        s.flush()
        data, index = s.output()
        s.close()
        st, raw, cl = index[0]
        self._merged = ifile.read_stream(data[st: st + cl]) if cl else []
        return self

    def get_reader(self):
        """KeyValuesReader: iterate (key_content, values list) groups."""
        deser_k = ifile.deserialize_text if self._text_keys \
            else ifile.deserialize_bytes_writable
        groups = []
        prev_key = None
        for k, v, same in self._merged:
            if not same and (prev_key is None or k != prev_key):
                groups.append((deser_k(k), []))
            groups[-1][1].append(v)
            prev_key = k
        return groups
