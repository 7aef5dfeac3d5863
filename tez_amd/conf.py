"""TezRuntimeConfiguration key mapping (same key names drive the engine —
SURVEY §2 'TezRuntimeConfiguration' row; defaults from
tez-runtime-library/.../api/TezRuntimeConfiguration.java):
  tez.runtime.io.sort.mb = 100                (:117)
  tez.runtime.io.sort.factor = 100            (:106)
  tez.runtime.key.class / value.class
  tez.runtime.key.comparator.class
  tez.runtime.empty.partitions.info-via-events.enabled = true
  tez.runtime.enable.final-merge.in.output = true
  tez.runtime.sorter.class = PIPELINED        (:170; both sorters map to the
                                               one GPU engine — SURVEY §2)
"""
from . import _engine


_KEY_CLASSES = {
    "org.apache.hadoop.io.BytesWritable": (_engine.KEY_BYTES, _engine.CMP_TEZBYTES),
    "org.apache.hadoop.io.Text": (_engine.KEY_TEXT, _engine.CMP_TEXT),
}

_COMPARATORS = {
    "org.apache.tez.runtime.library.common.comparator.TezBytesComparator":
        _engine.CMP_TEZBYTES,
    "org.apache.hadoop.io.Text$Comparator": _engine.CMP_TEXT,
}


def conf_from_tez_properties(props: dict, num_partitions: int):
    """Build a TzsConf from a flat tez.runtime.* key/value dict (the same keys
    an OrderedPartitionedKVEdgeConfig would carry)."""
    def b(key, dflt):
        v = props.get(key)
        if v is None:
            return dflt
        return str(v).lower() in ("1", "true", "yes")

    sorter_cls = props.get("tez.runtime.sorter.class", "PIPELINED")
    if str(sorter_cls).upper() not in ("PIPELINED", "LEGACY"):
        raise ValueError(f"unknown sorter.class {sorter_cls}")
    # PIPELINED and LEGACY (DefaultSorter) both map to the GPU engine
    # (one sorter serves both; TezRuntimeConfiguration.java:170)
    key_cls = props.get("tez.runtime.key.class",
                        "org.apache.hadoop.io.BytesWritable")
    if key_cls not in _KEY_CLASSES:
        raise ValueError(f"unsupported key class {key_cls} (round-1 engine "
                         "supports BytesWritable and Text)")
    key_type, cmp_default = _KEY_CLASSES[key_cls]
    cmp_cls = props.get("tez.runtime.key.comparator.class")
    if cmp_cls:
        if cmp_cls not in _COMPARATORS:
            # mirror the key-class handling: a silently-wrong sort order is
            # worse than an error
            raise ValueError(f"unsupported key.comparator.class {cmp_cls} "
                             "(engine supports TezBytesComparator and "
                             "Text.Comparator)")
        comparator = _COMPARATORS[cmp_cls]
    else:
        comparator = cmp_default

    combiner = 0
    comb_cls = props.get("tez.runtime.combiner.class")
    if comb_cls:
        val_cls = props.get("tez.runtime.value.class", "")
        if val_cls != "org.apache.hadoop.io.IntWritable":
            raise ValueError("round-1 combiner supports IntWritable-sum only "
                             "(SURVEY §8f row 1)")
        combiner = 1  # SUM_INT

    return _engine.make_conf(
        num_partitions,
        key_type=key_type,
        comparator=comparator,
        combiner=combiner,
        min_spills_for_combine=int(props.get("tez.runtime.combine.min.spills", 3)),
        send_empty_partition_details=int(
            b("tez.runtime.empty.partitions.info-via-events.enabled", True)),
        io_sort_factor=int(props.get("tez.runtime.io.sort.factor", 100)),
        final_merge_enabled=int(b("tez.runtime.enable.final-merge.in.output", True)),
        sort_buffer_bytes=int(props.get("tez.runtime.io.sort.mb", 100)) << 20,
    )
