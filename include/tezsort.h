/* tezsort.h — C-ABI of the MI355X-native ordered-shuffle engine (libtezsort.so).
 *
 * Drop-in boundary for apache/tez's ordered shuffle hot path.
 * Each entry point cites the reference interface it replaces
 * (paths under /root/reference, file:line).  A JNI shim binding these under
 * OrderedPartitionedKVOutput / OrderedGroupedKVInput is sketched in INTEGRATION.md.
 *
 * Conventions: all functions return 0 on success or a negative errno-style code;
 * tzs_last_error() returns a thread-local message for the last failure.
 * Pointers prefixed d_ are HIP device pointers; everything else is host memory.
 * Threading: each handle is used by one thread AT A TIME, but the thread may
 * change between calls (producer thread != flush thread — SURVEY §8b,
 * mirrors ExternalSorter threading, ExternalSorter.java:74-92 /
 * PipelinedSorter.java:399 "synchronized collect").  The device-buffer pool
 * and allocation registry behind every handle are mutex-protected, so
 * distinct handles may be driven from distinct threads concurrently; all
 * device work runs on the null HIP stream (serialized per process).
 */
#ifndef TEZSORT_H
#define TEZSORT_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- error handling ---------------------------------------------------- */
const char* tzs_last_error(void);

/* ---- configuration -----------------------------------------------------
 * Mirrors the TezRuntimeConfiguration keys that drive the ordered path
 * (TezRuntimeConfiguration.java:106-175,456-461). Key names map 1:1; the
 * Python layer (tez_amd/conf.py) parses the tez.runtime.* strings into this.
 */
typedef enum {
  TZS_KEY_BYTES = 0,   /* BytesWritable: 4B BE len + content (variable lengths supported; order = proxy-then-serialized) */
  TZS_KEY_TEXT  = 1    /* Text: vint len + UTF-8 content */
} tzs_key_type;

typedef enum {
  TZS_CMP_TEZBYTES = 0, /* TezBytesComparator.java:38-62 (memcmp over serialized form, 3-byte proxy) */
  TZS_CMP_TEXT     = 1  /* hadoop Text.Comparator: content memcmp, shorter first; no proxy */
} tzs_comparator;

typedef struct tzs_conf {
  int32_t num_partitions;              /* numOutputs */
  int32_t key_type;                    /* tzs_key_type */
  int32_t value_type;                  /* tzs_key_type (framing only) */
  int32_t comparator;                  /* tzs_comparator; tez.runtime.key.comparator.class */
  int32_t rle;                         /* -1 auto (DESIGN.md §3 gate), 0 off, 1 on */
  int32_t send_empty_partition_details;/* tez.runtime.empty.partitions.info-via-events.enabled, default 1 */
  int32_t io_sort_factor;              /* tez.runtime.io.sort.factor, default 100 (TezRuntimeConfiguration.java:106) */
  int32_t final_merge_enabled;         /* tez.runtime.enable.final-merge.in.output, default 1 */
  int64_t sort_buffer_bytes;           /* tez.runtime.io.sort.mb<<20; spill triggers when
                                          data bytes + 16*records exceed this */
  int32_t device;                      /* HIP device ordinal; -1 = current */
  int32_t world_size;                  /* GPUs sharing the shuffle (partition owner = p % world_size) */
  int32_t rank;                        /* this process's rank */
  int32_t combiner;                    /* 0 none; 1 SUM_INT: fold equal-key runs,
                                          summing 4B BE IntWritable values
                                          (runCombineProcessor call sites,
                                          PipelinedSorter.java:602-609,816-821) */
  int32_t min_spills_for_combine;      /* tez.runtime.combine.min.spills, default 3
                                          (PipelinedSorter.java:244) */
  int32_t discard_spill_streams;       /* 1: free each spill's emitted IFile bytes
                                          right after the spill completes (valid only
                                          with final merge enabled; the merge reads
                                          the columnar set — for workloads near the
                                          288 GB HBM capacity) */
} tzs_conf;

void tzs_conf_default(tzs_conf* c, int32_t num_partitions);

/* Index record triple — byte-compatible with TezIndexRecord.java:33-56. */
typedef struct tzs_index_record {
  int64_t start_offset;
  int64_t raw_length;    /* decompressedBytesWritten: IFile.java:396-418 */
  int64_t part_length;   /* bytes in file incl. header+CRC */
} tzs_index_record;

/* ---- map side: sorter ---------------------------------------------------
 * Replaces ExternalSorter/PipelinedSorter behind OrderedPartitionedKVOutput
 * (OrderedPartitionedKVOutput.java:150-219, PipelinedSorter.java:388-467,559-851).
 */
typedef struct tzs_sorter tzs_sorter;

int tzs_sorter_create(const tzs_conf* conf, tzs_sorter** out);

/* KeyValuesWriter.write(k, v) (OrderedPartitionedKVOutput.java:168-181).
 * key/val are SERIALIZED bytes (BytesWritable/Text form);
 * partition < 0 => computed via HashPartitioner semantics
 * (HashPartitioner.java:32-35). Host path: buffers and ships to device in batches. */
int tzs_sorter_write(tzs_sorter* s, const void* key, int32_t klen,
                     const void* val, int32_t vlen, int32_t partition);

/* Batched columnar device-resident variant (the bench/plugin fast path).
 * d_data: n records, record i = serialized key ‖ serialized value at
 * d_off[i] .. d_off[i+1]; d_klen[i] = serialized key length.
 * d_part: per-record partition (int32), or NULL => computed on device. */
int tzs_sorter_write_batch_device(tzs_sorter* s, const void* d_data,
                                  const uint64_t* d_off, const uint32_t* d_klen,
                                  const int32_t* d_part, int64_t n);

/* Zero-copy variant: the sorter takes OWNERSHIP of buffers previously
 * allocated by tzs_malloc_device / tzs_generate (the caller must not use or
 * free them afterwards).  Valid only as the first batch of a spill.  The
 * reference's collect() copies because input arrives record-at-a-time
 * (PipelinedSorter.java:399-467); device-resident producers hand whole
 * buffers over instead. */
int tzs_sorter_write_batch_device_adopt(tzs_sorter* s, void* d_data,
                                        uint64_t* d_off, uint32_t* d_klen,
                                        int32_t* d_part, int64_t n);

/* Force a spill of everything absorbed since the last spill
 * (PipelinedSorter.spill, PipelinedSorter.java:559-648). Returns spill id >= 0. */
int tzs_sorter_spill(tzs_sorter* s);

/* Reduce-side ingestion of an ALREADY-SORTED columnar segment (the
 * MergeManager admission path, MergeManager.java:423-519: fetched segments
 * arrive sorted by the map side).  Registers the caller's device buffers as
 * one sorted spill WITHOUT copying or re-sorting them (the caller keeps
 * them alive until flush/close); flush() then runs the k-way merge-path
 * merge over all segments (TezMerger.MergeQueue, TezMerger.java:466-706).
 * d_part = per-record original partition ids (explicit partitioners ride
 * through the exchange), or NULL to recompute HashPartitioner placement.
 * Returns the segment id (>=0) or <0 on error; -22 if the segment is not
 * sorted by (partition, key). */
int tzs_sorter_add_sorted_segment(tzs_sorter* s, const void* d_data,
                                  const uint64_t* d_off, const uint32_t* d_klen,
                                  const int32_t* d_part, int64_t n);

/* flush(): final spill + final merge (PipelinedSorter.flush, :665-851).
 * After this, final output/index are available. Blocking. */
int tzs_sorter_flush(tzs_sorter* s);

int tzs_sorter_num_spills(const tzs_sorter* s);

/* Compressed final output: the same partition segments
 * re-framed as TIF\1 — one zlib stream per segment (device deflate:
 * 32 KB chunks, fixed-Huffman LZ77 with stored fallback, sync-flush
 * stitching), CRC32 over the COMPRESSED payload, per the reference's
 * DefaultCodec framing (IFile.java:352-368; golden-fixture-pinned reader).
 * index: rawLength keeps the uncompressed accounting, partLength = 4 +
 * zlib stream + 4.  Buffer owned by the sorter (valid until close). */
int tzs_sorter_output_compressed(tzs_sorter* s, const void** d_bytes,
                                 int64_t* nbytes, tzs_index_record* index);

/* Final merged output: IFile bytes for all partitions concatenated (device
 * pointer, owned by the sorter) + index records (host). Valid until close. */
int tzs_sorter_output(tzs_sorter* s, const void** d_bytes, int64_t* nbytes,
                      tzs_index_record* index /* [num_partitions] */);

/* Per-spill segment view (for pipelined shuffle / exchange). */
int tzs_sorter_spill_output(tzs_sorter* s, int32_t spill_id,
                            const void** d_bytes, int64_t* nbytes,
                            tzs_index_record* index);

/* Materialize the reference on-disk layout (TezTaskOutputFiles.java:52-69):
 * <dir>/output/<unique_id>/file.out[.index], per-spill
 * <dir>/output/<unique_id>_<spill>/file.out[.index] when final merge is off. */
int tzs_sorter_write_files(tzs_sorter* s, const char* local_dir, const char* unique_id);

/* Final sort as permuted columnar arrays — the xGMI exchange wire
 * (DESIGN.md §4): records in sorted order, partition p's records at
 * rec_ranges[p]..rec_ranges[p+1] / bytes byte_ranges[p]..byte_ranges[p+1].
 * Pointers are device memory owned by the sorter (valid until close).
 * rec_ranges/byte_ranges are caller host arrays of num_partitions+1. */
int tzs_sorter_sorted_columnar(tzs_sorter* s, const void** d_data,
                               const uint64_t** d_off, const uint32_t** d_klen,
                               uint64_t* rec_ranges, uint64_t* byte_ranges);

/* Counters mirrored from TaskCounter semantics (ExternalSorter.java:217-225). */
typedef struct tzs_counters {
  int64_t output_records;   /* OUTPUT_RECORDS */
  int64_t output_bytes;     /* OUTPUT_BYTES: sum serialized k+v bytes (PipelinedSorter.java:466) */
  int64_t output_bytes_with_overhead; /* rawLength total */
  int64_t spilled_records;
  int64_t num_spills;
  int64_t rle_applied;      /* engine rule, DESIGN.md §3 */
} tzs_counters;
int tzs_sorter_counters(const tzs_sorter* s, tzs_counters* out);

void tzs_sorter_close(tzs_sorter* s);

/* ---- reduce side: merge --------------------------------------------------
 * Replaces MergeManager/TezMerger behind OrderedGroupedKVInput
 * (MergeManager.java:423-519,1162-1328; TezMerger.java:466-1066).
 * Segments are device-resident columnar record sets (the xGMI exchange wire,
 * DESIGN.md §4) or IFile-framed bytes.
 */
typedef struct tzs_segment {
  const void*     d_data;   /* columnar records (key‖val serialized) */
  const uint64_t* d_off;    /* [n+1] */
  const uint32_t* d_klen;   /* [n] */
  int64_t         n;
} tzs_segment;

/* Merge k sorted segments of ONE partition into an IFile stream
 * (TezMerger.MergeQueue.merge + writeFile, TezMerger.java:707-931,216-246).
 * Emits RLE/SAME_KEY runs per IFile.java:590-615 when rle. */
int tzs_merge_segments(const tzs_conf* conf, const tzs_segment* segs, int32_t nsegs,
                       void** d_out, int64_t* out_bytes, tzs_index_record* rec);

/* ---- synthetic input generation (bench/tests; device-resident) ---------- */
/* Seeded deterministic generator of serialized KV records in HBM
 * (shapes of BASELINE.json configs). kind: 0 = C2 (fixed klen random unique
 * content, fixed vlen), 1 = C3 (Text keys len in [klen_min,klen_max], Zipf word
 * + unique suffix), 2 = C5 (TeraSort 10B key + 90B value). */
int tzs_generate(uint64_t seed, int64_t n, int32_t kind,
                 int32_t klen, int32_t vlen, const tzs_conf* conf,
                 void** d_data, uint64_t** d_off, uint32_t** d_klen, int32_t** d_part);
void tzs_free_device(void* d_ptr);

/* Device-pool telemetry: out = {in-use bytes, held bytes, peak in-use bytes,
 * drop-and-retry count}.  A non-zero drop count means the working set
 * approached HBM capacity and the pool discarded idle buffers (hipMalloc
 * churn follows — a performance smell at large configs). */
void tzs_pool_stats(uint64_t out[4]);

/* ---- introspection ------------------------------------------------------ */
/* Per-phase HIP-event times of the last flush, nanoseconds. */
typedef struct tzs_times {
  int64_t absorb_ns, composite_ns, sort_ns, permute_ns, emit_ns, crc_ns, dominant_kernel_elems, total_ns;
  int64_t sort_passes;        /* radix passes actually run (incl. refinement) */
  int64_t dominant_kernel_ns; /* scatter total */
  int64_t merge_ns;           /* flush merge-path tree (k-way spill merge) */
} tzs_times;
int tzs_sorter_times(const tzs_sorter* s, tzs_times* out);

#ifdef __cplusplus
}
#endif
#endif /* TEZSORT_H */
