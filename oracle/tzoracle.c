#define _GNU_SOURCE /* qsort_r */
/* tzoracle.c — ORACLE: CPU restatement of apache/tez's ordered-shuffle hot path.
 *
 * TEST INFRASTRUCTURE ONLY.  This library is the parity yardstick for the GPU
 * engine (tez_amd).  Only tests/, __graft_entry__.smoke() (as the checker) and
 * bench.py's cpu_baseline leg may link or call it.  The product path must
 * never route through this code.
 *
 * Every function cites the reference it restates (paths under /root/reference,
 * apache/tez @ 2026-08-21, file:line).  Third-party boundary: hadoop-common
 * 3.5.0 (pom.xml:84) supplies WritableUtils vint, CRC32 (DataChecksum/
 * PureJavaCrc32 = ISO-HDLC), WritableComparator.hashBytes and the
 * Text/BytesWritable serialization; those are restated from their published
 * algorithms and pinned by the reference's golden fixture
 * (TestIFile_concatenated_compressed.bin) plus hand-computed vectors in tests
 * (SURVEY.md §8c).  Duplicate-(partition,key) tie order is UNPINNED vs the
 * reference (hadoop QuickSort is unstable): this oracle is deterministic
 * (stable by original index); bit-exact parity claims hold on unique-key
 * inputs (SURVEY.md §8c mitigation).
 */
#include <stdint.h>
#include <stdlib.h>
#include <string.h>
#include <stdio.h>
#include <pthread.h>

#define TZO_API __attribute__((visibility("default")))

/* ================= vint codec =================
 * Restates hadoop WritableUtils.writeVLong/readVLong (used at IFile.java:
 * 576-577,599 and ShuffleHeader.java:82-106): one byte for [-112,127]; else a
 * marker byte -113..-120 (positive, 1..8 BE bytes) / -121..-128 (negative,
 * value stored bitwise-NOT). */
TZO_API int tzo_vint_size(int64_t i) {
  if (i >= -112 && i <= 127) return 1;
  if (i < 0) i = ~i;
  int n = 0;
  while (i != 0) { i = (int64_t)((uint64_t)i >> 8); n++; }
  return n + 1;
}

TZO_API int tzo_vint_write(uint8_t* b, int64_t i) {
  if (i >= -112 && i <= 127) { b[0] = (uint8_t)i; return 1; }
  int len = -112;
  if (i < 0) { i = ~i; len = -120; }
  int64_t tmp = i;
  while (tmp != 0) { tmp = (int64_t)((uint64_t)tmp >> 8); len--; }
  b[0] = (uint8_t)len;
  int n = (len < -120) ? -(len + 120) : -(len + 112);
  for (int idx = n; idx != 0; idx--) {
    int shift = (idx - 1) * 8;
    b[n - idx + 1] = (uint8_t)((uint64_t)i >> shift);
  }
  return n + 1;
}

static int vint_decode_size(int8_t first) {
  if (first >= -112) return 1;
  if (first < -120) return -119 - first;
  return -111 - first;
}

TZO_API int tzo_vint_read(const uint8_t* b, int64_t* out) {
  int8_t first = (int8_t)b[0];
  int len = vint_decode_size(first);
  if (len == 1) { *out = first; return 1; }
  int64_t i = 0;
  for (int k = 1; k < len; k++) i = (i << 8) | b[k];
  int neg = (first < -120);
  *out = neg ? ~i : i;
  return len;
}

/* ================= CRC32 (ISO-HDLC, zlib-compatible) =================
 * Restates hadoop DataChecksum.Type.CRC32 (IFileOutputStream.java:55-57) and
 * PureJavaCrc32 (TezSpillRecord.java:65,123) — both the standard reflected
 * CRC-32 with polynomial 0xEDB88320, init/xorout 0xFFFFFFFF.  Semantics match
 * zlib's crc32(); tests cross-check against zlib. */
static uint32_t crc_table[256];
static int crc_table_init_done = 0;
static void crc_table_init(void) {
  if (crc_table_init_done) return;
  for (uint32_t n = 0; n < 256; n++) {
    uint32_t c = n;
    for (int k = 0; k < 8; k++) c = (c & 1) ? (0xEDB88320u ^ (c >> 1)) : (c >> 1);
    crc_table[n] = c;
  }
  crc_table_init_done = 1;
}

TZO_API uint32_t tzo_crc32(uint32_t crc, const uint8_t* p, size_t n) {
  crc_table_init();
  crc ^= 0xFFFFFFFFu;
  for (size_t i = 0; i < n; i++) crc = crc_table[(crc ^ p[i]) & 0xFF] ^ (crc >> 8);
  return crc ^ 0xFFFFFFFFu;
}

/* ================= java hash / partitioner =================
 * hashBytes: h = 1; h = 31*h + (signed)byte  (hadoop WritableComparator.
 * hashBytes; BytesWritable/Text hashCode via BinaryComparable).
 * partition = (hash & Integer.MAX_VALUE) % numPartitions
 * (partitioner/HashPartitioner.java:32-35). */
TZO_API int32_t tzo_hash_bytes(const uint8_t* p, int32_t n) {
  int32_t h = 1;
  for (int32_t i = 0; i < n; i++) h = (int32_t)((uint32_t)h * 31u) + (int8_t)p[i];
  return h;
}

TZO_API int32_t tzo_partition(const uint8_t* content, int32_t len, int32_t nparts) {
  return (tzo_hash_bytes(content, len) & 0x7fffffff) % nparts;
}

/* ================= comparators =================
 * id 0 = TezBytesComparator.java:38-42: unsigned memcmp over the FULL
 *        serialized key (incl. the 4B BE length prefix), then length.
 *        (WritableComparator.compareBytes semantics.)
 * id 1 = hadoop Text.Comparator: skip the vint length, unsigned memcmp of the
 *        UTF-8 content, shorter-first on ties. */
static int cmp_bytes_range(const uint8_t* a, int32_t la, const uint8_t* b, int32_t lb) {
  int32_t m = la < lb ? la : lb;
  int c = memcmp(a, b, (size_t)m);
  if (c != 0) return c < 0 ? -1 : 1;
  return (la < lb) ? -1 : (la > lb ? 1 : 0);
}

TZO_API int tzo_compare_key(int comparator, const uint8_t* a, int32_t la,
                            const uint8_t* b, int32_t lb) {
  if (comparator == 1) { /* Text */
    int na = vint_decode_size((int8_t)a[0]);
    int nb = vint_decode_size((int8_t)b[0]);
    return cmp_bytes_range(a + na, la - na, b + nb, lb - nb);
  }
  return cmp_bytes_range(a, la, b, lb); /* TezBytes: serialized form */
}

/* Key CONTENT view (strip the serialization prefix) — for hashing/proxy. */
TZO_API int tzo_key_content(int key_type, const uint8_t* k, int32_t klen,
                            const uint8_t** content, int32_t* clen) {
  if (key_type == 1) { /* Text: vint + utf8 */
    int n = vint_decode_size((int8_t)k[0]);
    *content = k + n; *clen = klen - n;
  } else {             /* BytesWritable: 4B BE + content */
    *content = k + 4; *clen = klen - 4;
  }
  return 0;
}

/* 3-byte proxy of TezBytesComparator.getProxy (TezBytesComparator.java:45-62):
 * first up to 3 CONTENT bytes packed (b0<<16 | b1<<8 | b2). */
static int32_t proxy3(const uint8_t* content, int32_t clen) {
  int32_t b1 = clen > 0 ? content[0] : 0;
  int32_t b2 = clen > 1 ? content[1] : 0;
  int32_t b3 = clen > 2 ? content[2] : 0;
  return (b1 << 16) | (b2 << 8) | b3;
}

/* bitcount(n): PipelinedSorter.java:317-324; partitionBits = bitcount(P)+1
 * (:165); prefix = (partition << (32-pb)) | (proxy >>> pb) (:451-457). */
static int bitcount(int n) { int b = 0; while (n != 0) { b++; n >>= 1; } return b; }

TZO_API uint32_t tzo_prefix(int comparator, int key_type, int32_t partition,
                            int32_t num_partitions, const uint8_t* key, int32_t klen) {
  int pb = bitcount(num_partitions) + 1;
  int32_t proxy = 0;
  if (comparator == 0) { /* TezBytesComparator implements ProxyComparator */
    const uint8_t* c; int32_t cl;
    tzo_key_content(key_type, key, klen, &c, &cl);
    proxy = proxy3(c, cl);
  }
  return ((uint32_t)partition << (32 - pb)) | ((uint32_t)proxy >> pb);
}

/* ================= growable byte buffer ================= */
typedef struct { uint8_t* p; size_t len, cap; } buf_t;
static void buf_reserve(buf_t* b, size_t need) {
  if (b->len + need <= b->cap) return;
  size_t nc = b->cap ? b->cap : 4096;
  while (nc < b->len + need) nc *= 2;
  b->p = (uint8_t*)realloc(b->p, nc);
  b->cap = nc;
}
static void buf_put(buf_t* b, const void* src, size_t n) {
  buf_reserve(b, n); memcpy(b->p + b->len, src, n); b->len += n;
}
static void buf_put_vint(buf_t* b, int64_t v, int64_t* raw) {
  uint8_t tmp[10]; int n = tzo_vint_write(tmp, v); buf_put(b, tmp, (size_t)n);
  if (raw) *raw += n;
}

TZO_API void tzo_free(void* p) { free(p); }

/* ================= IFile writer =================
 * Restates IFile.Writer (IFile.java:263-635), uncompressed:
 *  - header 'T','I','F',0 written to the raw stream BEFORE the checksum
 *    stream wraps it (IFile.java:337-339,374-380) => CRC covers payload only;
 *  - records {vint klen, vint vlen, key, val} (writeKVPair :573-588);
 *  - RLE: writeRLE marker -2 when a repeat starts (:590-603), V_END -3 when a
 *    run closes (:605-615); repeated values as {vint vlen, val} (:560-571);
 *  - close: V_END if needed, EOF {-1,-1}, CRC32 trailer big-endian
 *    (IFileOutputStream.java:81-90); rawLength accounting per :396-418. */
typedef struct tzo_writer {
  buf_t out;
  int rle;
  int prev_is_repeat;     /* prevKey == REPEAT_KEY */
  buf_t prevkey;          /* serialized previous key (valid when rle) */
  int64_t raw_len;        /* decompressedBytesWritten */
  int64_t nrec;
  int closed;
} tzo_writer;

TZO_API tzo_writer* tzo_writer_new(int rle) {
  tzo_writer* w = (tzo_writer*)calloc(1, sizeof(tzo_writer));
  w->rle = rle;
  static const uint8_t HDR[4] = { 'T', 'I', 'F', 0 };
  buf_put(&w->out, HDR, 4);
  return w;
}

/* append(DataInputBuffer,DataInputBuffer): IFile.java:535-558 */
TZO_API int tzo_writer_append(tzo_writer* w, const uint8_t* key, int32_t klen,
                              const uint8_t* val, int32_t vlen) {
  int same = 0;
  if (w->rle && klen != 0 && w->prevkey.len == (size_t)klen &&
      memcmp(w->prevkey.p, key, (size_t)klen) == 0)
    same = 1;
  if (!same) {
    if (w->prev_is_repeat) buf_put_vint(&w->out, -3, &w->raw_len); /* V_END */
    buf_put_vint(&w->out, klen, &w->raw_len);
    buf_put_vint(&w->out, vlen, &w->raw_len);
    buf_put(&w->out, key, (size_t)klen);
    buf_put(&w->out, val, (size_t)vlen);
    w->raw_len += klen + vlen;
    if (w->rle) { w->prevkey.len = 0; buf_put(&w->prevkey, key, (size_t)klen); }
  } else {
    if (!w->prev_is_repeat) buf_put_vint(&w->out, -2, &w->raw_len); /* RLE */
    buf_put_vint(&w->out, vlen, &w->raw_len);
    buf_put(&w->out, val, (size_t)vlen);
    w->raw_len += vlen;
  }
  w->prev_is_repeat = same;
  w->nrec++;
  return 0;
}

/* append(REPEAT_KEY, value): IFile.java:535-558 with key == REPEAT_KEY
 * (writeFile's SAME_KEY branch, TezMerger.java:222-228). */
TZO_API int tzo_writer_append_same(tzo_writer* w, const uint8_t* val, int32_t vlen) {
  if (!w->prev_is_repeat) buf_put_vint(&w->out, -2, &w->raw_len);
  buf_put_vint(&w->out, vlen, &w->raw_len);
  buf_put(&w->out, val, (size_t)vlen);
  w->raw_len += vlen;
  w->prev_is_repeat = 1;
  w->nrec++;
  return 0;
}

/* close(): IFile.java:382-436.  Returns (malloc'd) stream bytes. */
TZO_API int tzo_writer_close(tzo_writer* w, uint8_t** out, int64_t* out_len,
                             int64_t* raw_len, int64_t* part_len) {
  if (!w->closed) {
    if (w->prev_is_repeat) buf_put_vint(&w->out, -3, &w->raw_len);
    buf_put_vint(&w->out, -1, &w->raw_len);
    buf_put_vint(&w->out, -1, &w->raw_len);
    w->raw_len += 4; /* header bytes, IFile.java:402-403 */
    uint32_t crc = tzo_crc32(0, w->out.p + 4, w->out.len - 4);
    uint8_t t[4] = { (uint8_t)(crc >> 24), (uint8_t)(crc >> 16),
                     (uint8_t)(crc >> 8), (uint8_t)crc };
    buf_put(&w->out, t, 4);
    w->closed = 1;
  }
  if (raw_len) *raw_len = w->raw_len;
  if (part_len) *part_len = (int64_t)w->out.len;
  if (out) { *out = w->out.p; *out_len = (int64_t)w->out.len; w->out.p = NULL; }
  return 0;
}

TZO_API void tzo_writer_free(tzo_writer* w) {
  if (!w) return;
  free(w->out.p); free(w->prevkey.p); free(w);
}

/* ================= IFile reader =================
 * Restates IFile.Reader.positionToNextRecord/readRawKey/nextRawValue
 * (IFile.java:877-1001) over an in-memory stream.  Input may be a full stream
 * (header+payload+CRC: with_header=1, CRC verified) or a fetched in-memory
 * segment (payload+EOF only, header stripped, CRC consumed — SURVEY §8a a9:
 * with_header=0). Outputs flat arrays; SAME_KEY records get the stored key. */
typedef struct {
  int64_t n;            /* records */
  uint8_t* keys;        /* concatenated key bytes per record */
  int64_t* key_off;     /* [n+1] */
  uint8_t* vals;
  int64_t* val_off;     /* [n+1] */
  uint8_t* same_key;    /* [n] 1 if stream encoded it as SAME_KEY */
} tzo_records;

TZO_API int tzo_ifile_read(const uint8_t* stream, int64_t len, int with_header,
                           tzo_records** out) {
  const uint8_t* p = stream;
  const uint8_t* end;
  if (with_header) {
    if (len < 8) return -1;
    if (!(p[0] == 'T' && p[1] == 'I' && p[2] == 'F')) return -2;
    if (p[3] != 0) return -3; /* compressed unsupported in oracle reader */
    uint32_t want = ((uint32_t)stream[len - 4] << 24) | ((uint32_t)stream[len - 3] << 16) |
                    ((uint32_t)stream[len - 2] << 8) | (uint32_t)stream[len - 1];
    uint32_t got = tzo_crc32(0, stream + 4, (size_t)(len - 8));
    if (want != got) return -4;
    p += 4; end = stream + len - 4;
  } else {
    end = stream + len;
  }
  buf_t keys = {0}, vals = {0};
  buf_t koff = {0}, voff = {0}, same = {0};
  int64_t z = 0;
  buf_put(&koff, &z, 8); buf_put(&voff, &z, 8);
  buf_t curkey = {0};
  int prev_rle = 0;      /* currentKeyLength == RLE_MARKER */
  int rc = 0;
  int64_t n = 0;
  while (1) {
    int64_t klen, vlen;
    int is_same = 0;
    if (p >= end) { rc = -5; break; }
    if (prev_rle) { /* readValueLength, IFile.java:877-883 */
      p += tzo_vint_read(p, &vlen);
      if (vlen == -3) { /* V_END: fresh key+value lengths */
        p += tzo_vint_read(p, &klen);
        p += tzo_vint_read(p, &vlen);
      } else {
        klen = -2;
      }
    } else {
      p += tzo_vint_read(p, &klen);
      p += tzo_vint_read(p, &vlen);
    }
    if (klen == -1 && vlen == -1) break; /* EOF */
    if (klen == -2) {
      is_same = 1;
    } else {
      if (klen < 0 || vlen < 0 || p + klen > end) { rc = -6; break; }
      curkey.len = 0; buf_put(&curkey, p, (size_t)klen);
      p += klen;
    }
    if (vlen < 0 || p + vlen > end) { rc = -7; break; }
    buf_put(&keys, curkey.p, curkey.len);
    int64_t ko = (int64_t)keys.len; buf_put(&koff, &ko, 8);
    buf_put(&vals, p, (size_t)vlen);
    int64_t vo = (int64_t)vals.len; buf_put(&voff, &vo, 8);
    uint8_t s8 = (uint8_t)is_same; buf_put(&same, &s8, 1);
    p += vlen;
    prev_rle = (klen == -2) || (is_same == 0 && 0);
    prev_rle = (klen == -2);
    n++;
  }
  free(curkey.p);
  if (rc != 0) {
    free(keys.p); free(vals.p); free(koff.p); free(voff.p); free(same.p);
    return rc;
  }
  tzo_records* r = (tzo_records*)calloc(1, sizeof(tzo_records));
  r->n = n;
  r->keys = keys.p; r->key_off = (int64_t*)koff.p;
  r->vals = vals.p; r->val_off = (int64_t*)voff.p;
  r->same_key = same.p;
  *out = r;
  return 0;
}

TZO_API int64_t tzo_records_n(const tzo_records* r) { return r->n; }
TZO_API const uint8_t* tzo_records_keys(const tzo_records* r) { return r->keys; }
TZO_API const int64_t* tzo_records_key_off(const tzo_records* r) { return r->key_off; }
TZO_API const uint8_t* tzo_records_vals(const tzo_records* r) { return r->vals; }
TZO_API const int64_t* tzo_records_val_off(const tzo_records* r) { return r->val_off; }
TZO_API const uint8_t* tzo_records_same(const tzo_records* r) { return r->same_key; }
TZO_API void tzo_records_free(tzo_records* r) {
  if (!r) return;
  free(r->keys); free(r->key_off); free(r->vals); free(r->val_off); free(r->same_key);
  free(r);
}

/* ================= spill index =================
 * TezSpillRecord.writeToFile (TezSpillRecord.java:122-147): P × 3 BE longs
 * (startOffset, rawLength, partLength) + 8B BE PureJavaCrc32 of those bytes. */
static void put_be64(uint8_t* p, uint64_t v) {
  for (int i = 0; i < 8; i++) p[i] = (uint8_t)(v >> (56 - 8 * i));
}

TZO_API int tzo_index_encode(const int64_t* triples /* [3P] */, int32_t P,
                             uint8_t* out /* [24P+8] */) {
  for (int32_t i = 0; i < 3 * P; i++) put_be64(out + 8 * i, (uint64_t)triples[i]);
  uint32_t crc = tzo_crc32(0, out, (size_t)(24 * P));
  put_be64(out + 24 * P, (uint64_t)crc);
  return 24 * P + 8;
}

/* ================= map-side spill oracle =================
 * Restates one PipelinedSorter buffer-generation spill
 * (PipelinedSorter.java:399-467 collect, :966-1023 sort order, :559-648 spill):
 * records sorted by (prefix int, comparator over serialized key, original
 * index), then per partition appended to an rle-flagged IFile writer; empty
 * partitions get no writer when send_empty (spill :586-601).
 *
 * rle_mode: -1 = auto (adjacent-equal pairs > 0.1*n — engine rule, DESIGN.md
 * §3; the reference gate PipelinedSorter.java:1437-1439 is trace-dependent and
 * unpinnable), 0/1 forced.  Returns the data-file bytes, the index-file bytes,
 * and (optionally) the sorted record order. */
typedef struct {
  const uint8_t* data;
  const uint64_t* off;
  const uint32_t* klen;
  const int32_t* part;
  uint32_t* prefix;
  int comparator;
} sortctx_t;

static sortctx_t* g_ctx; /* single-threaded tzo_spill path */

static int spill_cmp_r(const void* A, const void* B, void* arg) {
  const sortctx_t* ctx = (const sortctx_t*)arg;
  int64_t ia = *(const int64_t*)A, ib = *(const int64_t*)B;
  uint32_t pa = ctx->prefix[ia], pb = ctx->prefix[ib];
  if (pa != pb) return pa < pb ? -1 : 1;
  const uint8_t* ka = ctx->data + ctx->off[ia];
  const uint8_t* kb = ctx->data + ctx->off[ib];
  int c = tzo_compare_key(ctx->comparator, ka, (int32_t)ctx->klen[ia],
                          kb, (int32_t)ctx->klen[ib]);
  if (c != 0) return c;
  return ia < ib ? -1 : (ia > ib ? 1 : 0);
}

static int spill_cmp(const void* A, const void* B) {
  int64_t ia = *(const int64_t*)A, ib = *(const int64_t*)B;
  uint32_t pa = g_ctx->prefix[ia], pb = g_ctx->prefix[ib];
  /* prefix ints compared as (signed) java ints: kvip - kvjp,
     PipelinedSorter.java:1014-1023; both non-negative by construction. */
  if (pa != pb) return pa < pb ? -1 : 1;
  const uint8_t* ka = g_ctx->data + g_ctx->off[ia];
  const uint8_t* kb = g_ctx->data + g_ctx->off[ib];
  int c = tzo_compare_key(g_ctx->comparator, ka, (int32_t)g_ctx->klen[ia],
                          kb, (int32_t)g_ctx->klen[ib]);
  if (c != 0) return c;
  return ia < ib ? -1 : (ia > ib ? 1 : 0); /* deterministic; unpinned vs reference */
}

/* Combiner restatement (runCombineProcessor call sites:
 * PipelinedSorter.java:602-609,816-821): fold runs of comparator-equal keys
 * within a partition, summing 4-byte big-endian IntWritable values (two's
 * complement wrap, as java int addition does).  combiner: 0 = none,
 * 1 = SUM_INT. */
static int32_t be32(const uint8_t* p) {
  return (int32_t)(((uint32_t)p[0] << 24) | ((uint32_t)p[1] << 16) |
                   ((uint32_t)p[2] << 8) | (uint32_t)p[3]);
}
static void put_be32(uint8_t* p, int32_t v) {
  p[0] = (uint8_t)((uint32_t)v >> 24); p[1] = (uint8_t)((uint32_t)v >> 16);
  p[2] = (uint8_t)((uint32_t)v >> 8); p[3] = (uint8_t)v;
}

TZO_API int tzo_spill(
    /* records: record i = serialized key ‖ serialized value at off[i]..off[i+1] */
    const uint8_t* data, const uint64_t* off, const uint32_t* klen,
    const int32_t* part_in /* may be NULL => HashPartitioner */,
    int64_t n,
    int32_t P, int key_type, int comparator, int rle_mode, int send_empty,
    int combiner,
    uint8_t** out_data, int64_t* out_data_len,
    uint8_t** out_index, int64_t* out_index_len,
    int64_t* out_order /* [n] or NULL */, int* out_rle) {
  int32_t* part = (int32_t*)malloc(sizeof(int32_t) * (size_t)(n ? n : 1));
  uint32_t* prefix = (uint32_t*)malloc(sizeof(uint32_t) * (size_t)(n ? n : 1));
  int64_t* order = (int64_t*)malloc(sizeof(int64_t) * (size_t)(n ? n : 1));
  for (int64_t i = 0; i < n; i++) {
    const uint8_t* k = data + off[i];
    if (part_in) part[i] = part_in[i];
    else {
      const uint8_t* c; int32_t cl;
      tzo_key_content(key_type, k, (int32_t)klen[i], &c, &cl);
      part[i] = tzo_partition(c, cl, P);
    }
    prefix[i] = tzo_prefix(comparator, key_type, part[i], P, k, (int32_t)klen[i]);
    order[i] = i;
  }
  sortctx_t ctx = { data, off, klen, part, prefix, comparator };
  g_ctx = &ctx;
  qsort(order, (size_t)n, sizeof(int64_t), spill_cmp);
  g_ctx = NULL;

  int rle = rle_mode;
  if (rle_mode < 0) {
    int64_t eq = 0;
    for (int64_t i = 1; i < n; i++) {
      int64_t a = order[i - 1], b = order[i];
      if (part[a] == part[b] &&
          tzo_compare_key(comparator, data + off[a], (int32_t)klen[a],
                          data + off[b], (int32_t)klen[b]) == 0)
        eq++;
    }
    rle = (eq * 10 > n) ? 1 : 0; /* eq > 0.1*total */
  }

  buf_t file = {0};
  int64_t* triples = (int64_t*)calloc((size_t)(3 * P), sizeof(int64_t));
  int64_t pos = 0;
  /* records are (partition, …)-sorted: walk runs */
  for (int32_t p = 0; p < P; p++) {
    /* find run of this partition */
    int64_t lo = pos;
    while (pos < n && part[order[pos]] == p) pos++;
    int has = (pos > lo);
    int64_t start = (int64_t)file.len;
    int64_t rawl = 0, partl = 0;
    if (has || !send_empty) {
      tzo_writer* w = tzo_writer_new(rle);
      if (combiner == 1) {
        /* fold equal-key runs, emit (key, sum) */
        int64_t i = lo;
        while (i < pos) {
          int64_t r0 = order[i];
          const uint8_t* kb = data + off[r0];
          int32_t kl = (int32_t)klen[r0];
          int32_t sum = 0;
          int64_t j = i;
          for (; j < pos; j++) {
            int64_t r = order[j];
            if (j > i && tzo_compare_key(comparator, kb, kl, data + off[r],
                                         (int32_t)klen[r]) != 0)
              break;
            sum = (int32_t)((uint32_t)sum +
                            (uint32_t)be32(data + off[r] + klen[r]));
          }
          uint8_t vb[4];
          put_be32(vb, sum);
          tzo_writer_append(w, kb, kl, vb, 4);
          i = j;
        }
      } else {
        for (int64_t i = lo; i < pos; i++) {
          int64_t r = order[i];
          const uint8_t* kb = data + off[r];
          const uint8_t* vb = kb + klen[r];
          int32_t vl = (int32_t)(off[r + 1] - off[r] - klen[r]);
          tzo_writer_append(w, kb, (int32_t)klen[r], vb, vl);
        }
      }
      uint8_t* seg; int64_t seglen;
      tzo_writer_close(w, &seg, &seglen, &rawl, &partl);
      tzo_writer_free(w);
      buf_put(&file, seg, (size_t)seglen);
      free(seg);
    }
    triples[3 * p + 0] = start;
    triples[3 * p + 1] = rawl;
    triples[3 * p + 2] = partl;
  }
  uint8_t* idx = (uint8_t*)malloc((size_t)(24 * P + 8));
  tzo_index_encode(triples, P, idx);
  free(triples);
  if (out_order) memcpy(out_order, order, sizeof(int64_t) * (size_t)n);
  free(order); free(prefix); free(part);
  *out_data = file.p; *out_data_len = (int64_t)file.len;
  *out_index = idx; *out_index_len = 24 * P + 8;
  if (out_rle) *out_rle = rle;
  return 0;
}

/* ---- partition-parallel spill (the "fair" multi-core CPU baseline,
 * BASELINE.md): identical bytes to tzo_spill, with per-partition quicksort +
 * IFile emission fanned across nthreads.  Restates the same reference
 * semantics; parallelism is by partition (the reference's own unit of
 * independence). */
typedef struct {
  sortctx_t* ctx;
  int64_t* order;          /* full order array, partition-bucketed */
  const int64_t* pstart;   /* [P+1] partition record ranges */
  int32_t P;
  int rle;
  int key_combiner;
  int p_next;              /* shared work counter */
  pthread_mutex_t mu;
  uint8_t** seg;           /* per-partition emitted stream (or NULL) */
  int64_t* seg_len;
  int64_t* seg_raw;
  int send_empty;
} mtspill_t;

static void* mtspill_worker(void* vp) {
  mtspill_t* W = (mtspill_t*)vp;
  for (;;) {
    pthread_mutex_lock(&W->mu);
    int p = W->p_next++;
    pthread_mutex_unlock(&W->mu);
    if (p >= W->P) return NULL;
    int64_t lo = W->pstart[p], hi = W->pstart[p + 1];
    qsort_r(W->order + lo, (size_t)(hi - lo), sizeof(int64_t), spill_cmp_r, W->ctx);
    if (hi == lo && W->send_empty) { W->seg[p] = NULL; continue; }
    tzo_writer* w = tzo_writer_new(W->rle);
    const sortctx_t* ctx = W->ctx;
    for (int64_t i = lo; i < hi; i++) {
      int64_t r = W->order[i];
      const uint8_t* kb = ctx->data + ctx->off[r];
      const uint8_t* vb = kb + ctx->klen[r];
      int32_t vl = (int32_t)(ctx->off[r + 1] - ctx->off[r] - ctx->klen[r]);
      tzo_writer_append(w, kb, (int32_t)ctx->klen[r], vb, vl);
    }
    int64_t seglen;
    tzo_writer_close(w, &W->seg[p], &seglen, &W->seg_raw[p], &W->seg_len[p]);
    tzo_writer_free(w);
  }
}

/* Multithreaded variant of tzo_spill (rle auto only over unique-key use is
 * the baseline case; rle_mode is honored the same way).  combiner
 * unsupported here (baseline measures the sort+emit path). */
TZO_API int tzo_spill_mt(
    const uint8_t* data, const uint64_t* off, const uint32_t* klen,
    const int32_t* part_in, int64_t n,
    int32_t P, int key_type, int comparator, int rle_mode, int send_empty,
    int nthreads,
    uint8_t** out_data, int64_t* out_data_len,
    uint8_t** out_index, int64_t* out_index_len) {
  int32_t* part = (int32_t*)malloc(sizeof(int32_t) * (size_t)(n ? n : 1));
  uint32_t* prefix = (uint32_t*)malloc(sizeof(uint32_t) * (size_t)(n ? n : 1));
  for (int64_t i = 0; i < n; i++) {
    const uint8_t* k = data + off[i];
    if (part_in) part[i] = part_in[i];
    else {
      const uint8_t* c; int32_t cl;
      tzo_key_content(key_type, k, (int32_t)klen[i], &c, &cl);
      part[i] = tzo_partition(c, cl, P);
    }
    prefix[i] = tzo_prefix(comparator, key_type, part[i], P, k, (int32_t)klen[i]);
  }
  /* bucket records by partition, stable by index */
  int64_t* pstart = (int64_t*)calloc((size_t)(P + 2), sizeof(int64_t));
  for (int64_t i = 0; i < n; i++) pstart[part[i] + 1]++;
  for (int32_t p = 0; p < P; p++) pstart[p + 1] += pstart[p];
  int64_t* order = (int64_t*)malloc(sizeof(int64_t) * (size_t)(n ? n : 1));
  int64_t* cur = (int64_t*)malloc(sizeof(int64_t) * (size_t)(P ? P : 1));
  memcpy(cur, pstart, sizeof(int64_t) * (size_t)P);
  for (int64_t i = 0; i < n; i++) order[cur[part[i]]++] = i;
  free(cur);

  sortctx_t ctx = { data, off, klen, part, prefix, comparator };
  int rle = rle_mode;
  mtspill_t W;
  W.ctx = &ctx; W.order = order; W.pstart = pstart; W.P = P;
  W.send_empty = send_empty; W.p_next = 0;
  W.key_combiner = 0;
  pthread_mutex_init(&W.mu, NULL);
  W.seg = (uint8_t**)calloc((size_t)P, sizeof(uint8_t*));
  W.seg_len = (int64_t*)calloc((size_t)P, sizeof(int64_t));
  W.seg_raw = (int64_t*)calloc((size_t)P, sizeof(int64_t));
  if (rle < 0) rle = 0; /* auto gate needs global adjacency; baseline inputs
                           are unique-key (C2 shape) => off, like tzo_spill */
  W.rle = rle;
  if (nthreads < 1) nthreads = 1;
  pthread_t th[64];
  if (nthreads > 64) nthreads = 64;
  for (int t = 0; t < nthreads; t++) pthread_create(&th[t], NULL, mtspill_worker, &W);
  for (int t = 0; t < nthreads; t++) pthread_join(th[t], NULL);
  pthread_mutex_destroy(&W.mu);

  buf_t file = {0};
  int64_t* triples = (int64_t*)calloc((size_t)(3 * P), sizeof(int64_t));
  for (int32_t p = 0; p < P; p++) {
    triples[3 * p + 0] = (int64_t)file.len;
    if (W.seg[p]) {
      triples[3 * p + 1] = W.seg_raw[p];
      triples[3 * p + 2] = W.seg_len[p];
      buf_put(&file, W.seg[p], (size_t)W.seg_len[p]);
      free(W.seg[p]);
    }
  }
  uint8_t* idx = (uint8_t*)malloc((size_t)(24 * P + 8));
  tzo_index_encode(triples, P, idx);
  free(triples); free(W.seg); free(W.seg_len); free(W.seg_raw);
  free(order); free(pstart); free(prefix); free(part);
  *out_data = file.p; *out_data_len = (int64_t)file.len;
  *out_index = idx; *out_index_len = 24 * P + 8;
  return 0;
}

/* ================= k-way merge oracle (TezMerger restatement) ============
 * Restates TezMerger.MergeQueue for one partition's segments
 * (TezMerger.java:466-706 heap + SAME_KEY, :707-931 multi-pass with
 * getPassFactor, :216-246 writeFile REPEAT_KEY emission).
 *
 * Heap: binary heap ordered by comparator on current keys (lessThan :696-705);
 * ties broken by segment insertion order (deterministic; unpinned vs the
 * reference's hadoop PriorityQueue — irrelevant for unique keys).
 * SAME_KEY state machine: adjustPriorityQueue/compareKeyWithNextTopKey
 * (:598-653): within-segment RLE (reader SAME_KEY) keeps the segment on top;
 * a new key or segment end triggers a cross-segment prev-key comparison. */
typedef struct {
  tzo_records* recs;
  int64_t pos;        /* next record index */
  int64_t partlen;    /* segment length for the length sort */
  int seq;            /* insertion order for tie-break */
} mseg_t;

typedef struct {
  mseg_t** heap; int size;
  int comparator;
} mheap_t;

static int seg_key_cmp(int comparator, mseg_t* a, mseg_t* b) {
  const uint8_t* ka = a->recs->keys + a->recs->key_off[a->pos];
  int32_t la = (int32_t)(a->recs->key_off[a->pos + 1] - a->recs->key_off[a->pos]);
  const uint8_t* kb = b->recs->keys + b->recs->key_off[b->pos];
  int32_t lb = (int32_t)(b->recs->key_off[b->pos + 1] - b->recs->key_off[b->pos]);
  int c = tzo_compare_key(comparator, ka, la, kb, lb);
  if (c != 0) return c;
  return a->seq - b->seq;
}
static void heap_down(mheap_t* h, int i) {
  for (;;) {
    int l = 2 * i + 1, r = l + 1, m = i;
    if (l < h->size && seg_key_cmp(h->comparator, h->heap[l], h->heap[m]) < 0) m = l;
    if (r < h->size && seg_key_cmp(h->comparator, h->heap[r], h->heap[m]) < 0) m = r;
    if (m == i) break;
    mseg_t* t = h->heap[i]; h->heap[i] = h->heap[m]; h->heap[m] = t; i = m;
  }
}

/* Merge segments (already parsed to records) of ONE partition into an IFile
 * stream.  rle_mode as in tzo_spill (auto = adjacent-equal rule over the
 * merged sequence).  segments with 0 records are dropped (merge :780-789). */
static int merge_records(mseg_t** segs, int nsegs, int comparator, int rle_mode,
                         int emit_empty /* create writer when no records */,
                         uint8_t** out, int64_t* out_len,
                         int64_t* rawl, int64_t* partl) {
  mheap_t h = { (mseg_t**)malloc(sizeof(mseg_t*) * (size_t)(nsegs ? nsegs : 1)), 0, comparator };
  for (int i = 0; i < nsegs; i++) {
    if (segs[i]->recs->n > 0) { h.heap[h.size++] = segs[i]; }
  }
  for (int i = h.size / 2 - 1; i >= 0; i--) heap_down(&h, i);
  if (h.size == 0 && !emit_empty) {
    free(h.heap);
    *out = NULL; *out_len = 0; *rawl = 0; *partl = 0;
    return 0;
  }
  /* First pass: produce the merged (segment,record) sequence + same_key flags
     per the MergeQueue state machine, so an auto-RLE decision can precede
     emission. */
  int64_t total = 0;
  for (int i = 0; i < h.size; i++) total += h.heap[i]->recs->n;
  int32_t* seq_seg = (int32_t*)malloc(sizeof(int32_t) * (size_t)(total ? total : 1));
  int64_t* seq_rec = (int64_t*)malloc(sizeof(int64_t) * (size_t)(total ? total : 1));
  uint8_t* seq_same = (uint8_t*)malloc((size_t)(total ? total : 1));
  int64_t m = 0;
  /* prev key for cross-segment SAME_KEY (compareKeyWithNextTopKey :642-653) */
  buf_t prevkey = {0};
  int have_prev = 0;
  int same_state = 0; /* hasNext == SAME_KEY */
  while (h.size > 0) {
    mseg_t* top = h.heap[0];
    tzo_records* R = top->recs;
    int64_t i = top->pos;
    seq_seg[m] = top->seq; seq_rec[m] = i; seq_same[m] = (uint8_t)same_state;
    m++;
    /* store prev key BEFORE advancing unless the state was SAME_KEY
       (adjustPriorityQueue :598-621) */
    if (!same_state || !have_prev) {
      prevkey.len = 0;
      buf_put(&prevkey, R->keys + R->key_off[i],
              (size_t)(R->key_off[i + 1] - R->key_off[i]));
      have_prev = 1;
    }
    top->pos++;
    if (top->pos >= R->n) {
      /* segment exhausted: pop, then cross-segment prev-key compare */
      h.heap[0] = h.heap[--h.size];
      if (h.size) heap_down(&h, 0);
      same_state = 0;
      if (h.size > 0) {
        mseg_t* nt = h.heap[0];
        const uint8_t* nk = nt->recs->keys + nt->recs->key_off[nt->pos];
        int32_t nl = (int32_t)(nt->recs->key_off[nt->pos + 1] - nt->recs->key_off[nt->pos]);
        if (tzo_compare_key(comparator, nk, nl, prevkey.p, (int32_t)prevkey.len) == 0)
          same_state = 1;
      }
    } else if (R->same_key[top->pos]) {
      /* within-segment RLE: do not rebalance (adjustPriorityQueue :633-635) */
      same_state = 1;
    } else {
      heap_down(&h, 0);
      same_state = 0;
      mseg_t* nt = h.heap[0];
      if (nt != top) {
        const uint8_t* nk = nt->recs->keys + nt->recs->key_off[nt->pos];
        int32_t nl = (int32_t)(nt->recs->key_off[nt->pos + 1] - nt->recs->key_off[nt->pos]);
        if (tzo_compare_key(comparator, nk, nl, prevkey.p, (int32_t)prevkey.len) == 0)
          same_state = 1;
      }
    }
  }
  free(prevkey.p);
  int rle = rle_mode;
  if (rle_mode < 0) {
    int64_t eq = 0;
    for (int64_t i = 1; i < m; i++) if (seq_same[i]) eq++;
    rle = (eq * 10 > m) ? 1 : 0;
  }
  tzo_writer* w = tzo_writer_new(rle);
  /* writeFile (TezMerger.java:216-246): SAME_KEY => append(REPEAT_KEY, v) */
  for (int64_t i = 0; i < m; i++) {
    tzo_records* R = segs[seq_seg[i]]->recs;
    int64_t r = seq_rec[i];
    const uint8_t* vb = R->vals + R->val_off[r];
    int32_t vl = (int32_t)(R->val_off[r + 1] - R->val_off[r]);
    if (seq_same[i]) {
      tzo_writer_append_same(w, vb, vl);
    } else {
      const uint8_t* kb = R->keys + R->key_off[r];
      int32_t kl = (int32_t)(R->key_off[r + 1] - R->key_off[r]);
      tzo_writer_append(w, kb, kl, vb, vl);
    }
  }
  tzo_writer_close(w, out, out_len, rawl, partl);
  tzo_writer_free(w);
  free(seq_seg); free(seq_rec); free(seq_same); free(h.heap);
  return 0;
}

/* Final merge of numSpills spill files into one output file + index
 * (PipelinedSorter.flush, PipelinedSorter.java:759-851): per partition,
 * TezMerger.merge over the spill segments (single pass when k <= factor;
 * multi-pass restatement of getPassFactor :921-931 applies intermediate
 * merges to in-memory temp segments).  Writer rle per rle_mode. */
TZO_API int tzo_final_merge(
    const uint8_t* const* spill_data, const int64_t* spill_len,
    const uint8_t* const* spill_index /* 24P+8 each; big-endian triples */,
    int32_t nspills, int32_t P, int comparator, int rle_mode, int send_empty,
    int32_t factor, int combiner /* 0 none, 1 SUM_INT (gated by caller) */,
    uint8_t** out_data, int64_t* out_data_len,
    uint8_t** out_index, int64_t* out_index_len) {
  (void)spill_len;
  buf_t file = {0};
  int64_t* triples = (int64_t*)calloc((size_t)(3 * P), sizeof(int64_t));
  int rc = 0;
  for (int32_t p = 0; p < P && rc == 0; p++) {
    /* collect this partition's segments from each spill index */
    mseg_t** segs = (mseg_t**)calloc((size_t)nspills, sizeof(mseg_t*));
    int ns = 0;
    int should_write = 0;
    for (int32_t s = 0; s < nspills; s++) {
      const uint8_t* ix = spill_index[s] + 24 * p;
      int64_t start = 0, rawl = 0, partl = 0;
      for (int i = 0; i < 8; i++) start = (start << 8) | ix[i];
      for (int i = 8; i < 16; i++) rawl = (rawl << 8) | ix[i];
      for (int i = 16; i < 24; i++) partl = (partl << 8) | ix[i];
      /* hasData: rawLength > HEADER+2 (TezIndexRecord.java:52-56) */
      if (rawl > 6 || !send_empty) {
        should_write = 1;
        if (partl > 0) {
          tzo_records* R = NULL;
          int r2 = tzo_ifile_read(spill_data[s] + start, partl, 1, &R);
          if (r2 != 0) { rc = r2; break; }
          mseg_t* ms = (mseg_t*)calloc(1, sizeof(mseg_t));
          ms->recs = R; ms->pos = 0; ms->partlen = partl; ms->seq = ns;
          segs[ns++] = ms;
        }
      }
    }
    int64_t start = (int64_t)file.len;
    int64_t rawl = 0, partl = 0;
    if (rc == 0) {
      /* sortSegments when k > factor (PipelinedSorter.java:796-797): stable
         by length.  Multi-pass when k > factor (TezMerger merge :753-913). */
      if (ns > factor) {
        /* stable sort by partlen (insertion sort; ns is small) */
        for (int i = 1; i < ns; i++) {
          mseg_t* x = segs[i]; int j = i - 1;
          while (j >= 0 && segs[j]->partlen > x->partlen) { segs[j + 1] = segs[j]; j--; }
          segs[j + 1] = x;
        }
        int passno = 1;
        while (ns > factor) {
          int f = factor;
          if (passno == 1) {
            int mod = (ns - 1) % (factor - 1);
            if (mod != 0) f = mod + 1;
          }
          /* merge first f segments into a temp segment */
          for (int i = 0; i < f; i++) segs[i]->seq = i;
          uint8_t* tb; int64_t tl, trl, tpl;
          merge_records(segs, f, comparator, /*rle*/ 0, 1, &tb, &tl, &trl, &tpl);
          for (int i = 0; i < f; i++) { tzo_records_free(segs[i]->recs); free(segs[i]); }
          tzo_records* TR = NULL;
          tzo_ifile_read(tb, tl, 1, &TR);
          free(tb);
          mseg_t* ms = (mseg_t*)calloc(1, sizeof(mseg_t));
          ms->recs = TR; ms->pos = 0; ms->partlen = tpl;
          /* insert into remaining list keeping length order (binarySearch
             insert, TezMerger.java:886-893) */
          int rem = ns - f;
          memmove(segs, segs + f, sizeof(mseg_t*) * (size_t)rem);
          int ins = rem;
          for (int i = 0; i < rem; i++) if (segs[i]->partlen >= tpl) { ins = i; break; }
          memmove(segs + ins + 1, segs + ins, sizeof(mseg_t*) * (size_t)(rem - ins));
          segs[ins] = ms;
          ns = rem + 1;
          passno++;
        }
      }
      for (int i = 0; i < ns; i++) segs[i]->seq = i;
      if (should_write) {
        uint8_t* seg; int64_t seglen;
        if (combiner == 1) {
          /* fold across the merged stream (runCombineProcessor at final
             merge, PipelinedSorter.java:816-821): flatten the merge, fold,
             re-emit */
          uint8_t* tmp; int64_t tl, trl, tpl;
          merge_records(segs, ns, comparator, 0, 1, &tmp, &tl, &trl, &tpl);
          tzo_records* R = NULL;
          tzo_ifile_read(tmp, tl, 1, &R);
          free(tmp);
          tzo_writer* w = tzo_writer_new(rle_mode == 1);
          int64_t i = 0;
          while (i < R->n) {
            const uint8_t* kb = R->keys + R->key_off[i];
            int32_t kl = (int32_t)(R->key_off[i + 1] - R->key_off[i]);
            int32_t sum = 0;
            int64_t j = i;
            for (; j < R->n; j++) {
              const uint8_t* kb2 = R->keys + R->key_off[j];
              int32_t kl2 = (int32_t)(R->key_off[j + 1] - R->key_off[j]);
              if (j > i && tzo_compare_key(comparator, kb, kl, kb2, kl2) != 0)
                break;
              sum = (int32_t)((uint32_t)sum +
                              (uint32_t)be32(R->vals + R->val_off[j]));
            }
            uint8_t vb[4];
            put_be32(vb, sum);
            tzo_writer_append(w, kb, kl, vb, 4);
            i = j;
          }
          tzo_writer_close(w, &seg, &seglen, &rawl, &partl);
          tzo_writer_free(w);
          tzo_records_free(R);
        } else {
          merge_records(segs, ns, comparator, rle_mode, 1, &seg, &seglen, &rawl, &partl);
        }
        if (seg) { buf_put(&file, seg, (size_t)seglen); free(seg); }
      }
    }
    for (int i = 0; i < ns; i++) { tzo_records_free(segs[i]->recs); free(segs[i]); }
    free(segs);
    triples[3 * p + 0] = start;
    triples[3 * p + 1] = rawl;
    triples[3 * p + 2] = partl;
  }
  if (rc != 0) { free(file.p); free(triples); return rc; }
  uint8_t* idx = (uint8_t*)malloc((size_t)(24 * P + 8));
  tzo_index_encode(triples, P, idx);
  free(triples);
  *out_data = file.p; *out_data_len = (int64_t)file.len;
  *out_index = idx; *out_index_len = 24 * P + 8;
  return 0;
}

/* ================= ShuffleHeader codec =================
 * ShuffleHeader.java:82-106: {vint idlen, idbytes, vlong clen, vlong rlen,
 * vint partition}. */
TZO_API int tzo_shuffle_header_write(uint8_t* out, const char* map_id,
                                     int64_t clen, int64_t rlen, int32_t partition) {
  int n = 0;
  int idlen = (int)strlen(map_id);
  n += tzo_vint_write(out + n, idlen);
  memcpy(out + n, map_id, (size_t)idlen); n += idlen;
  n += tzo_vint_write(out + n, clen);
  n += tzo_vint_write(out + n, rlen);
  n += tzo_vint_write(out + n, partition);
  return n;
}

TZO_API int tzo_shuffle_header_read(const uint8_t* in, char* map_id, int map_id_cap,
                                    int64_t* clen, int64_t* rlen, int32_t* partition) {
  int n = 0; int64_t v;
  n += tzo_vint_read(in + n, &v);
  int idlen = (int)v;
  if (idlen + 1 > map_id_cap) return -1;
  memcpy(map_id, in + n, (size_t)idlen); map_id[idlen] = 0; n += idlen;
  n += tzo_vint_read(in + n, &v); *clen = v;
  n += tzo_vint_read(in + n, &v); *rlen = v;
  n += tzo_vint_read(in + n, &v); *partition = (int32_t)v;
  return n;
}
