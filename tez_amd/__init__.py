"""tez_amd — MI355X-native rebuild of apache/tez's ordered-shuffle hot path.

Product package: the compute path is hand-written HIP (gfx950) behind the
C-ABI in include/tezsort.h; this package mirrors the reference's
OrderedPartitionedKVOutput / OrderedGroupedKVInput plugin surface
(DESIGN.md §1).  No CPU fallback: a missing native engine raises.
"""
from ._engine import (  # noqa: F401
    KEY_BYTES, KEY_TEXT, CMP_TEZBYTES, CMP_TEXT,
    Sorter, make_conf, upload_records, generate, free_device, device_available,
    pool_stats,
    merge_segments, read_device,
)
from .conf import conf_from_tez_properties  # noqa: F401
