// tezsort.hip — MI355X-native (gfx950/CDNA4) ordered-shuffle engine.
//
// From-scratch rebuild of apache/tez's shuffle sort/merge hot path
// (tez-runtime-library PipelinedSorter/TezMerger/IFile — see DESIGN.md and the
// reference citations in include/tezsort.h).  NOT a port: the map-side
// quicksort+spill and reduce-side heap merge are replaced by a stable
// segmented LSD radix sort over (partition, key-content) composites with
// refinement levels, device-side IFile emission and GF(2)-combined CRC32 —
// all HBM-bandwidth-bound byte work (no MFMA; north star).
//
// PRODUCT PATH.  The CPU oracle under /root/repo/oracle is test-only and is
// never called from here.
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>
#include <mutex>
#include <algorithm>
#include <sys/stat.h>
#include <sys/types.h>

#include "tezsort.h"

#define WAVE 64
#define BLOCK 256
#define WPB (BLOCK / WAVE)
#define RADIX 256
#define TILE_ROUNDS 12
#define TILE (TILE_ROUNDS * BLOCK) /* elements per scatter block.
  Round-1 (wavehist ranking): 16 rounds at 256 threads measured WORSE
  (doubled LDS halved resident blocks).  Round-2 barrier-light ranking
  unions the count region with the key staging, so 12 rounds at 1024
  threads keeps 16 waves/CU while growing digit runs to ~576 B. */

/* ------------------------------------------------------------------ */
/* error plumbing                                                      */
/* ------------------------------------------------------------------ */
static thread_local char g_err[1024];
extern "C" const char* tzs_last_error(void) { return g_err; }

#define FAIL(code, ...)                                        \
  do {                                                         \
    snprintf(g_err, sizeof(g_err), __VA_ARGS__);               \
    return (code);                                             \
  } while (0)

#define HIP_CHECK(x)                                                        \
  do {                                                                      \
    hipError_t _e = (x);                                                    \
    if (_e != hipSuccess)                                                   \
      FAIL(-70, "%s:%d HIP error %d (%s) in %s", __FILE__, __LINE__, (int)_e, \
           hipGetErrorString(_e), #x);                                      \
  } while (0)

/* ------------------------------------------------------------------ */
/* small host helpers                                                  */
/* ------------------------------------------------------------------ */
static uint32_t h_crc_table[256];
static bool h_crc_init_done = false;
static void h_crc_init() {
  if (h_crc_init_done) return;
  for (uint32_t n = 0; n < 256; n++) {
    uint32_t c = n;
    for (int k = 0; k < 8; k++) c = (c & 1) ? (0xEDB88320u ^ (c >> 1)) : (c >> 1);
    h_crc_table[n] = c;
  }
  h_crc_init_done = true;
}
static uint32_t h_crc32(uint32_t crc, const uint8_t* p, size_t n) {
  h_crc_init();
  crc ^= 0xFFFFFFFFu;
  for (size_t i = 0; i < n; i++) crc = h_crc_table[(crc ^ p[i]) & 0xFF] ^ (crc >> 8);
  return crc ^ 0xFFFFFFFFu;
}

/* GF(2) CRC-combine matrices (zlib crc32_combine algorithm restated).
 * mat[k] (32 u32 columns) applies 2^k ZERO BYTES to a raw crc register.
 * combine(c1, c2, len2) = apply(len2 bytes of zeros to c1) ^ c2. */
static uint32_t gf2_times(const uint32_t* m, uint32_t v) {
  uint32_t s = 0;
  for (int i = 0; v; i++, v >>= 1)
    if (v & 1) s ^= m[i];
  return s;
}
static void gf2_square(uint32_t* sq, const uint32_t* m) {
  for (int i = 0; i < 32; i++) sq[i] = gf2_times(m, m[i]);
}
#define CRC_MATS 40
static uint32_t h_crc_shift_mat[CRC_MATS][32]; /* [k]: 2^k zero bytes */
static bool h_mats_done = false;
static void h_build_crc_mats() {
  if (h_mats_done) return;
  uint32_t odd[32], even[32];
  odd[0] = 0xEDB88320u; /* one zero BIT operator (reflected poly) */
  for (int i = 1; i < 32; i++) odd[i] = 1u << (i - 1);
  /* square three times: 1 bit -> 2 -> 4 -> 8 bits = 1 byte */
  gf2_square(even, odd);
  gf2_square(odd, even);
  gf2_square(even, odd); /* even = 1 zero byte */
  memcpy(h_crc_shift_mat[0], even, sizeof(even));
  for (int k = 1; k < CRC_MATS; k++)
    gf2_square(h_crc_shift_mat[k], h_crc_shift_mat[k - 1]);
  h_mats_done = true;
}
static uint32_t h_crc_shift(uint32_t crc, uint64_t nbytes) {
  h_build_crc_mats();
  for (int k = 0; nbytes; k++, nbytes >>= 1)
    if (nbytes & 1) crc = gf2_times(h_crc_shift_mat[k], crc);
  return crc;
}
/* combine with zlib-style pre/post conditioning cancellation:
 * crc(A||B) = shift(crc(A)^0xff.., |B|) ^ (crc(B)^0xff..) ... the standard
 * identity: crc_combine(c1,c2,n) = shift_raw(c1, n) ^ c2 where shift_raw
 * operates on the plain value.  We validate against composition in tests. */
static uint32_t h_crc_combine(uint32_t c1, uint32_t c2, uint64_t len2) {
  return h_crc_shift(c1, len2) ^ c2;
}
extern "C" uint32_t tzs_test_crc_combine(uint32_t c1, uint32_t c2, uint64_t len2) {
  return h_crc_combine(c1, c2, len2);
}
extern "C" uint32_t tzs_test_crc32(uint32_t c, const void* p, uint64_t n) {
  return h_crc32(c, (const uint8_t*)p, n);
}

/* hadoop vint (WritableUtils) — host side for tiny patches */
static int h_vint_size(int64_t i) {
  if (i >= -112 && i <= 127) return 1;
  if (i < 0) i = ~i;
  int n = 0;
  while (i != 0) { i = (int64_t)((uint64_t)i >> 8); n++; }
  return n + 1;
}

/* ------------------------------------------------------------------ */
/* device helpers                                                      */
/* ------------------------------------------------------------------ */
__device__ __forceinline__ int d_vint_size(int64_t i) {
  if (i >= -112 && i <= 127) return 1;
  uint64_t u = (i < 0) ? ~(uint64_t)i : (uint64_t)i;
  int n = 0;
  while (u != 0) { u >>= 8; n++; }
  return n + 1;
}
__device__ __forceinline__ int d_vint_write(uint8_t* b, int64_t i) {
  if (i >= -112 && i <= 127) { b[0] = (uint8_t)i; return 1; }
  int len = -112;
  if (i < 0) { i = ~i; len = -120; }
  int64_t tmp = i;
  while (tmp != 0) { tmp = (int64_t)((uint64_t)tmp >> 8); len--; }
  b[0] = (uint8_t)len;
  int n = (len < -120) ? -(len + 120) : -(len + 112);
  for (int idx = n; idx != 0; idx--)
    b[n - idx + 1] = (uint8_t)((uint64_t)i >> ((idx - 1) * 8));
  return n + 1;
}
/* vint decoded size from first byte (for Text content offset) */
__device__ __forceinline__ int d_vint_decoded_size(int8_t first) {
  if (first >= -112) return 1;
  if (first < -120) return -119 - first;
  return -111 - first;
}

__device__ __forceinline__ int32_t d_hash_bytes(const uint8_t* p, int32_t n) {
  /* h = 31*h + signed(byte) — fetch 8 bytes at a time (byte loads cost an
     instruction each and serialize the chain on memory latency) */
  if (n == 16) {
    /* split the 16-step chain into two independent 8-chains:
       h = A*31^8 + B with A seeded 1, B seeded 0 (exact same mod-2^32 value) */
    uint64_t w0, w1;
    __builtin_memcpy(&w0, p, 8);
    __builtin_memcpy(&w1, p + 8, 8);
    int32_t A = 1, B = 0;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      A = (int32_t)((uint32_t)A * 31u) + (int8_t)(uint8_t)(w0 >> (8 * j));
      B = (int32_t)((uint32_t)B * 31u) + (int8_t)(uint8_t)(w1 >> (8 * j));
    }
    uint32_t p8 = 1;
#pragma unroll
    for (int j = 0; j < 8; j++) p8 *= 31u;
    return (int32_t)((uint32_t)A * p8 + (uint32_t)B);
  }
  int32_t h = 1;
  int32_t i = 0;
  for (; i + 8 <= n; i += 8) {
    uint64_t w;
    __builtin_memcpy(&w, p + i, 8);
    for (int j = 0; j < 8; j++)
      h = (int32_t)((uint32_t)h * 31u) + (int8_t)(uint8_t)(w >> (8 * j));
  }
  for (; i < n; i++) h = (int32_t)((uint32_t)h * 31u) + (int8_t)p[i];
  return h;
}

/* Record table: the union of any number of record sets (spill segments).
 * Segment 0 is inlined in the kernarg (the dominant single-spill case pays
 * no indirection); segments >= 1 live in a small device descriptor array —
 * no kernarg k-bound, so a 199-segment reduce merge needs no coalesce copy
 * (TezMerger.java:921-931 handles any k; VERDICT r1 missing #2). */
struct SegDesc {
  const uint8_t* data;
  const uint64_t* off;
  const uint32_t* klen;
  /* uniform-record fast path: when every record of the segment has the same
     serialized size/klen (C2/C5 shapes), offsets become arithmetic and the
     off/klen gathers (PMC: 12.7 GB per descriptor pass at n=1e8) vanish. */
  uint32_t rec_u;   /* record bytes, 0 = non-uniform */
  uint32_t klen_u;
};
struct RecTable {
  const uint8_t* data0;
  const uint64_t* off0;
  const uint32_t* klen0;
  const SegDesc* segs;   /* device array [nspills]; unused when nspills==1 */
  const uint32_t* base;  /* device array [nspills+1]; unused when nspills==1 */
  uint32_t rec_u0, klen_u0;
  uint32_t n0;           /* records in segment 0 */
  int32_t nspills;
  int32_t key_type; /* 0 bytes, 1 text */
};

__device__ __forceinline__ int rt_spill_of(const RecTable& rt, uint32_t g) {
  if (rt.nspills == 1 || g < rt.n0) return 0;
  uint32_t lo = 1, hi = (uint32_t)rt.nspills; /* g >= base[1] == n0 */
  while (lo + 1 < hi) {
    uint32_t mid = (lo + hi) >> 1;
    if (rt.base[mid] <= g) lo = mid; else hi = mid;
  }
  return (int)lo;
}
struct RecView {
  const uint8_t* key;   /* serialized key */
  uint32_t klen;
  const uint8_t* val;   /* serialized value */
  uint32_t vlen;
  const uint8_t* content;
  uint32_t clen;
};
__device__ __forceinline__ RecView rt_view(const RecTable& rt, uint32_t g) {
  const uint8_t* data;
  const uint64_t* off;
  const uint32_t* klen;
  uint32_t rec_u, klen_u, r;
  if (rt.nspills == 1 || g < rt.n0) {
    data = rt.data0; off = rt.off0; klen = rt.klen0;
    rec_u = rt.rec_u0; klen_u = rt.klen_u0; r = g;
  } else {
    int s = rt_spill_of(rt, g);
    SegDesc sd = rt.segs[s];
    data = sd.data; off = sd.off; klen = sd.klen;
    rec_u = sd.rec_u; klen_u = sd.klen_u; r = g - rt.base[s];
  }
  RecView v;
  if (rec_u) {
    v.key = data + (uint64_t)r * rec_u;
    v.klen = klen_u;
    v.val = v.key + v.klen;
    v.vlen = rec_u - v.klen;
  } else {
    uint64_t o = off[r];
    v.key = data + o;
    v.klen = klen[r];
    v.val = v.key + v.klen;
    v.vlen = (uint32_t)(off[r + 1] - o - v.klen);
  }
  if (rt.key_type == 1) {
    int n = d_vint_decoded_size((int8_t)v.key[0]);
    v.content = v.key + n;
    v.clen = v.klen - n;
  } else {
    v.content = v.key + 4;
    v.clen = v.klen - 4;
  }
  return v;
}

/* forward declarations (definitions follow the host code below) */
__global__ void k_shift_offsets(const uint64_t* src, uint64_t* dst, uint64_t shift,
                                int64_t n);
__global__ void k_max_u32(const uint32_t* a, uint32_t n, uint32_t* out);
__global__ void k_part_hist(const uint32_t* parts, uint32_t n, uint32_t* counts, int P);
__global__ void k_gather_part_meta(const uint64_t* pstart, const uint64_t* scan,
                                   const uint8_t* same, uint64_t total_body,
                                   uint32_t n, int P, uint64_t* out);
__global__ void k_check_uniform(const uint64_t* off, const uint32_t* klen, int64_t n,
                                uint64_t* mm /* {min_diff,max_diff,min_klen,max_klen} */);

/* ---- partition + composite ---- */
__global__ void k_hash_partition(RecTable rt, int32_t P, int32_t* d_part, uint32_t n) {
  /* 2 records per iteration: the per-record serial hash chain left waves
     parked on one gather at a time (85% WAIT_ANY, r2 PMC) */
  uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += 2 * stride) {
    uint32_t j = i + stride;
    RecView v0 = rt_view(rt, i);
    int32_t h0 = d_hash_bytes(v0.content, (int32_t)v0.clen);
    if (j < n) {
      RecView v1 = rt_view(rt, j);
      int32_t h1 = d_hash_bytes(v1.content, (int32_t)v1.clen);
      d_part[j] = (h1 & 0x7fffffff) % P;
    }
    d_part[i] = (h0 & 0x7fffffff) % P;
  }
}

/* composite = (part << (64-pbits)) | (first content bytes BE >> pbits),
 * masked to the top sort_bytes bytes: the radix sort only orders those
 * (adaptive pass count — DESIGN.md §4); rarer-than-1% ties are resolved by
 * the refinement levels, whose equality test must see the same mask. */
__device__ __forceinline__ uint64_t d_composite_one(
    const RecTable& rt, const int32_t* d_part, int32_t P, int pbits,
    int ref_pb, int ser_mode, uint64_t mask, uint32_t i) {
  RecView v = rt_view(rt, i);
  uint32_t part;
  if (d_part) part = (uint32_t)d_part[i];
  else part = (uint32_t)((d_hash_bytes(v.content, (int32_t)v.clen) & 0x7fffffff) % P);
  uint64_t key;
  if (ser_mode) {
    uint32_t proxy = ((v.clen > 0 ? (uint32_t)v.content[0] : 0u) << 16) |
                     ((v.clen > 1 ? (uint32_t)v.content[1] : 0u) << 8) |
                     (v.clen > 2 ? (uint32_t)v.content[2] : 0u);
    int pw = 24 - ref_pb;
    if (pw < 0) pw = 0;
    uint32_t proxy_t = pw ? (proxy >> (24 - pw)) : 0;
    uint64_t ser = 0;
    uint32_t m = v.klen < 8 ? v.klen : 8;
    for (uint32_t b = 0; b < m; b++) ser |= (uint64_t)v.key[b] << (56 - 8 * b);
    key = ((uint64_t)part << (64 - pbits))
          | ((uint64_t)proxy_t << (64 - pbits - pw))
          | (ser >> (pbits + pw));
  } else {
    uint64_t c = 0;
    uint32_t m = v.clen < 8 ? v.clen : 8;
    for (uint32_t b = 0; b < m; b++) c |= (uint64_t)v.content[b] << (56 - 8 * b);
    key = pbits ? (((uint64_t)part << (64 - pbits)) | (c >> pbits)) : c;
  }
  return key & mask;
}

/* 2 records per iteration: independent hash/gather chains keep two record
 * reads in flight (the kernel measured 64-67% wave-parked) */
__global__ void k_build_composite2(RecTable rt, const int32_t* d_part, int32_t P,
                                   int pbits, int ref_pb, int sort_bytes,
                                   int ser_mode, uint64_t* d_key, uint32_t* d_idx,
                                   uint32_t n) {
  uint64_t mask = (sort_bytes >= 8) ? ~0ull : ~0ull << (8 * (8 - sort_bytes));
  uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += 4 * stride) {
    uint32_t i1 = i + stride, i2 = i + 2 * stride, i3 = i + 3 * stride;
    uint64_t k0 = d_composite_one(rt, d_part, P, pbits, ref_pb, ser_mode, mask, i);
    uint64_t k1 = (i1 < n) ? d_composite_one(rt, d_part, P, pbits, ref_pb,
                                             ser_mode, mask, i1) : 0;
    uint64_t k2 = (i2 < n) ? d_composite_one(rt, d_part, P, pbits, ref_pb,
                                             ser_mode, mask, i2) : 0;
    uint64_t k3 = (i3 < n) ? d_composite_one(rt, d_part, P, pbits, ref_pb,
                                             ser_mode, mask, i3) : 0;
    d_key[i] = k0;
    d_idx[i] = i;
    if (i1 < n) { d_key[i1] = k1; d_idx[i1] = i1; }
    if (i2 < n) { d_key[i2] = k2; d_idx[i2] = i2; }
    if (i3 < n) { d_key[i3] = k3; d_idx[i3] = i3; }
  }
}

__global__ void k_build_composite(RecTable rt, const int32_t* d_part, int32_t P,
                                  int pbits, int ref_pb, int sort_bytes, int ser_mode,
                                  uint64_t* d_key, uint32_t* d_idx, uint32_t n) {
  /* d_part == nullptr: compute the HashPartitioner placement here (fused —
     a separate hash kernel re-reads all content bytes, ~10 GB at n=1e8).
     ser_mode (TezBytesComparator): order = partition, 3-byte content proxy,
     then the SERIALIZED key bytes (4B BE length first => length-then-content
     ties) — the reference prefix-int + raw-compare order
     (PipelinedSorter.java:451-463, TezBytesComparator.java:38-62).
     !ser_mode (Text): partition then content bytes, shorter-first ties. */
  uint64_t mask = (sort_bytes >= 8) ? ~0ull : ~0ull << (8 * (8 - sort_bytes));
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x) {
    RecView v = rt_view(rt, i);
    uint32_t part;
    if (d_part) part = (uint32_t)d_part[i];
    else part = (uint32_t)((d_hash_bytes(v.content, (int32_t)v.clen) & 0x7fffffff) % P);
    uint64_t key;
    if (ser_mode) {
      /* the reference prefix keeps only proxy >>> (bitcount(P)+1) bits
         (PipelinedSorter.java:457): ties on the TRUNCATED proxy fall through
         to the serialized compare, so the composite must truncate too */
      uint32_t proxy = ((v.clen > 0 ? (uint32_t)v.content[0] : 0u) << 16) |
                       ((v.clen > 1 ? (uint32_t)v.content[1] : 0u) << 8) |
                       (v.clen > 2 ? (uint32_t)v.content[2] : 0u);
      int pw = 24 - ref_pb;            /* surviving proxy bits */
      if (pw < 0) pw = 0;
      uint32_t proxy_t = pw ? (proxy >> (24 - pw)) : 0;
      uint64_t ser = 0;
      uint32_t m = v.klen < 8 ? v.klen : 8;
      for (uint32_t b = 0; b < m; b++) ser |= (uint64_t)v.key[b] << (56 - 8 * b);
      key = ((uint64_t)part << (64 - pbits))
            | ((uint64_t)proxy_t << (64 - pbits - pw))
            | (ser >> (pbits + pw));
    } else {
      uint64_t c = 0;
      uint32_t m = v.clen < 8 ? v.clen : 8;
      for (uint32_t b = 0; b < m; b++) c |= (uint64_t)v.content[b] << (56 - 8 * b);
      key = pbits ? (((uint64_t)part << (64 - pbits)) | (c >> pbits)) : c;
    }
    d_key[i] = key & mask;
    d_idx[i] = i;
  }
}

/* ---- stable LSD radix pass (8-bit digit) ----
 * hist: per-block digit counts over its contiguous tile.
 * scan: off[b][d] = digit_base[d] + sum_{b'<b} cnt[b'][d]  (digit-major scan)
 * scatter: stable rank within tile via wave ballots + LDS wave histograms. */
template <typename KeyT, int BLK = BLOCK>
__global__ void k_radix_hist(const KeyT* keys, uint32_t n, int byte_idx,
                             uint32_t* counts /* [nblocks*RADIX] */) {
  constexpr uint32_t OT = (uint32_t)TILE_ROUNDS * BLK;
  __shared__ uint32_t h[RADIX];
  for (int i = threadIdx.x; i < RADIX; i += blockDim.x) h[i] = 0;
  __syncthreads();
  uint32_t start = blockIdx.x * OT;
  uint32_t end = min(start + OT, n);
  for (uint32_t i = start + threadIdx.x; i < end; i += blockDim.x) {
    uint32_t d = (uint32_t)(keys[i] >> (8 * byte_idx)) & 0xFF;
    atomicAdd(&h[d], 1u);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < RADIX; i += blockDim.x)
    counts[blockIdx.x * RADIX + i] = h[i];
}

/* one block per digit: exclusive scan over blocks, writes running offsets and
 * the digit total */
__global__ void k_radix_scan_blocks(const uint32_t* counts, uint32_t nblocks,
                                    uint32_t* offsets, uint32_t* totals) {
  int d = blockIdx.x;
  __shared__ uint32_t lds[BLOCK];
  uint32_t running = 0;
  for (uint32_t b0 = 0; b0 < nblocks; b0 += blockDim.x) {
    uint32_t b = b0 + threadIdx.x;
    uint32_t v = (b < nblocks) ? counts[b * RADIX + d] : 0;
    /* inclusive block scan (Hillis-Steele in LDS) */
    lds[threadIdx.x] = v;
    __syncthreads();
    for (int s = 1; s < BLOCK; s <<= 1) {
      uint32_t t = (threadIdx.x >= (uint32_t)s) ? lds[threadIdx.x - s] : 0;
      __syncthreads();
      lds[threadIdx.x] += t;
      __syncthreads();
    }
    if (b < nblocks) offsets[b * RADIX + d] = running + lds[threadIdx.x] - v;
    uint32_t chunk_total = lds[BLOCK - 1];
    running += chunk_total;
    __syncthreads();
  }
  if (threadIdx.x == 0) totals[d] = running;
}

__global__ void k_radix_scan_digits(uint32_t* totals, uint32_t* bases) {
  /* single block of 256: exclusive scan of totals into bases */
  __shared__ uint32_t lds[RADIX];
  int t = threadIdx.x;
  lds[t] = totals[t];
  __syncthreads();
  for (int s = 1; s < RADIX; s <<= 1) {
    uint32_t v = (t >= s) ? lds[t - s] : 0;
    __syncthreads();
    lds[t] += v;
    __syncthreads();
  }
  bases[t] = lds[t] - totals[t];
}

/* Stable scatter with up to two u32 payloads — barrier-light ranking.
 * Each (round, wave) pair owns a private u16 count slot, so the 8 ranking
 * rounds run with NO barriers (the round-1 version re-used one wavehist
 * row per round: ~3 barriers x rounds; PMC showed the scatters 61-71%
 * wave-parked on barrier/latency, not bandwidth).  One in-place per-digit
 * prefix over the (round,wave) slots then yields every element's stable
 * tile rank; digit offsets come from a single wave-shuffle scan.  The count
 * region is re-used for the key staging (consumed into registers first).
 * Stores stay digit-contiguous (partial-line scatter waste was the round-1
 * fix). */
template <typename KeyT, bool HAS_A1, bool HAS_B64 = false, int BLK = BLOCK>
__global__ __launch_bounds__(BLK) void k_radix_scatter(
    const KeyT* keys_in, KeyT* keys_out,
    const uint32_t* a0_in, uint32_t* a0_out,
    const uint32_t* a1_in, uint32_t* a1_out,
    const uint64_t* b64_in, uint64_t* b64_out,
    uint32_t n, int byte_idx,
    const uint32_t* offsets, const uint32_t* bases) {
  constexpr uint32_t OT = (uint32_t)TILE_ROUNDS * BLK;
  constexpr int OWPB = BLK / WAVE;
  constexpr int RW = OWPB * TILE_ROUNDS;
  constexpr size_t CNT_BYTES = (size_t)RW * RADIX * 2;
  constexpr size_t KEY_BYTES = (size_t)OT * sizeof(KeyT);
  constexpr size_t UNION_BYTES = CNT_BYTES > KEY_BYTES ? CNT_BYTES : KEY_BYTES;
  __shared__ __attribute__((aligned(16))) uint8_t u_raw[UNION_BYTES];
  __shared__ uint32_t ls_a0[OT];
  __shared__ uint32_t ls_a1[HAS_A1 ? OT : 1];
  __shared__ uint64_t ls_b64[HAS_B64 ? OT : 1];
  __shared__ uint32_t tileoff[RADIX];
  uint16_t* cnt = (uint16_t*)u_raw;
  KeyT* ls_key = (KeyT*)u_raw;
  for (uint32_t i = threadIdx.x; i < (uint32_t)RW * RADIX / 2; i += blockDim.x)
    ((uint32_t*)cnt)[i] = 0;
  __syncthreads();
  uint32_t start = blockIdx.x * OT;
  uint32_t end = min(start + OT, n);
  uint32_t count = (start < n) ? (end - start) : 0;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wv = threadIdx.x / WAVE;
  const uint64_t lt_mask = (lane == 0) ? 0ull : (~0ull >> (64 - lane));
  KeyT my_key[TILE_ROUNDS];
  uint32_t my_a0[TILE_ROUNDS];
  uint32_t my_a1v[HAS_A1 ? TILE_ROUNDS : 1];
  uint64_t my_b64[HAS_B64 ? TILE_ROUNDS : 1];
  uint16_t my_rank[TILE_ROUNDS];
  uint16_t my_dig[TILE_ROUNDS];
  /* load phase first: every round's gathers issue back-to-back so the
     memory latency is paid once per tile, not once per ranking round */
  #pragma unroll
  for (int r = 0; r < TILE_ROUNDS; r++) {
    uint32_t i = start + (uint32_t)r * BLK + threadIdx.x;
    bool active = i < end;
    my_key[r] = active ? keys_in[i] : (KeyT)0;
    my_a0[r] = active ? a0_in[i] : 0;
    if (HAS_A1) my_a1v[r] = active ? a1_in[i] : 0;
    if (HAS_B64) my_b64[r] = active ? b64_in[i] : 0;
  }
  for (int r = 0; r < TILE_ROUNDS; r++) {
    uint32_t i = start + (uint32_t)r * BLK + threadIdx.x;
    bool active = i < end;
    uint32_t d = active ? ((uint32_t)(my_key[r] >> (8 * byte_idx)) & 0xFF) : 0u;
    uint64_t m = ~0ull;
    for (int b = 0; b < 8; b++) {
      uint64_t bb = __ballot((d >> b) & 1);
      m &= ((d >> b) & 1) ? bb : ~bb;
    }
    uint64_t act = __ballot(active);
    m &= act;
    uint32_t lane_rank = (uint32_t)__popcll(m & lt_mask);
    if (active && lane_rank == 0)
      cnt[(r * OWPB + wv) * RADIX + d] = (uint16_t)__popcll(m);
    my_rank[r] = (uint16_t)lane_rank;
    my_dig[r] = (uint16_t)d;
  }
  __syncthreads();
  if (threadIdx.x < RADIX) {
    uint32_t d = threadIdx.x;
    uint32_t running = 0;
    for (int rw = 0; rw < RW; rw++) {
      uint32_t t = cnt[rw * RADIX + d];
      cnt[rw * RADIX + d] = (uint16_t)running;
      running += t;
    }
    tileoff[d] = running;
  }
  __syncthreads();
  if (wv == 0) {
    uint32_t v0 = tileoff[4 * lane], v1 = tileoff[4 * lane + 1],
             v2 = tileoff[4 * lane + 2], v3 = tileoff[4 * lane + 3];
    uint32_t sum = v0 + v1 + v2 + v3;
    uint32_t inc = sum;
    for (int s2 = 1; s2 < WAVE; s2 <<= 1) {
      uint32_t t = __shfl_up(inc, s2);
      if (lane >= s2) inc += t;
    }
    uint32_t base0 = inc - sum;
    tileoff[4 * lane] = base0;
    tileoff[4 * lane + 1] = base0 + v0;
    tileoff[4 * lane + 2] = base0 + v0 + v1;
    tileoff[4 * lane + 3] = base0 + v0 + v1 + v2;
  }
  __syncthreads();
  uint16_t my_slot[TILE_ROUNDS];
  for (int r = 0; r < TILE_ROUNDS; r++) {
    uint32_t d = my_dig[r];
    my_slot[r] = (uint16_t)(tileoff[d] + cnt[(r * OWPB + wv) * RADIX + d] +
                            my_rank[r]);
  }
  __syncthreads(); /* cnt consumed; region becomes the key staging */
  for (int r = 0; r < TILE_ROUNDS; r++) {
    uint32_t i = start + (uint32_t)r * BLK + threadIdx.x;
    if (i < end) {
      uint32_t slot = my_slot[r];
      ls_key[slot] = my_key[r];
      ls_a0[slot] = my_a0[r];
      if (HAS_A1) ls_a1[slot] = my_a1v[r];
      if (HAS_B64) ls_b64[slot] = my_b64[r];
    }
  }
  __syncthreads();
  for (uint32_t j = threadIdx.x; j < count; j += blockDim.x) {
    KeyT k = ls_key[j];
    uint32_t d = (uint32_t)(k >> (8 * byte_idx)) & 0xFF;
    uint32_t pos = offsets[blockIdx.x * RADIX + d] + bases[d] + (j - tileoff[d]);
    keys_out[pos] = k;
    a0_out[pos] = ls_a0[j];
    if (HAS_A1) a1_out[pos] = ls_a1[j];
    if (HAS_B64) b64_out[pos] = ls_b64[j];
  }
}

/* ---- onesweep radix pass (single-kernel, decoupled lookback) ----
 * Global per-digit totals are permutation-invariant, so ONE read of the keys
 * yields the histograms of EVERY pass; each pass is then a single scatter
 * kernel: tiles are taken in order via a global ticket (dispatch order is
 * undefined on CDNA4 — a blockIdx-ordered lookback could deadlock, a ticket
 * order cannot: every ticket below yours belongs to a block that has already
 * started), tile digit counts are published as single-word agent-scope
 * atomics with a 2-bit status packed in (the guide's data-is-the-flag R2
 * form), and each tile resolves its exclusive prefix by walking back until
 * an INCLUSIVE entry.  Spins are bounded; on timeout an error flag makes the
 * host redo the pass with the classic 3-kernel path. */
#define OS_AGG (1u << 30)
#define OS_INC (2u << 30)
#define OS_CNT_MASK ((1u << 30) - 1)
typedef unsigned int __attribute__((address_space(1))) os_gu32;

template <typename KeyT>
__global__ void k_global_hist_all(const KeyT* keys, uint32_t n, int first_byte,
                                  int npasses, uint32_t* counts /* [npasses][256] */) {
  extern __shared__ uint32_t lh[]; /* npasses * 256 */
  for (int i = threadIdx.x; i < npasses * RADIX; i += blockDim.x) lh[i] = 0;
  __syncthreads();
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x) {
    KeyT k = keys[i];
    for (int p = 0; p < npasses; p++)
      atomicAdd(&lh[p * RADIX + ((uint32_t)(k >> (8 * (first_byte + p))) & 0xFF)], 1u);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < npasses * RADIX; i += blockDim.x)
    if (lh[i]) atomicAdd(&counts[i], lh[i]);
}

/* bases[p][d] = exclusive prefix of counts[p][*] — one block per pass
 * (replaces a host D2H + prefix + H2D round trip per base sort) */
__global__ void k_os_bases(const uint32_t* counts, uint32_t* bases) {
  __shared__ uint32_t tot[RADIX];
  int p = blockIdx.x;
  int d = threadIdx.x;
  tot[d] = counts[p * RADIX + d];
  __syncthreads();
  if (d == 0) {
    uint32_t run = 0;
    for (int i = 0; i < RADIX; i++) {
      uint32_t t = tot[i];
      tot[i] = run;
      run += t;
    }
  }
  __syncthreads();
  bases[p * RADIX + d] = tot[d];
}

template <typename KeyT, bool HAS_A1, bool HAS_B64 = false, int BLK = BLOCK>
__global__ __launch_bounds__(BLK) void k_onesweep_pass(
    const KeyT* keys_in, KeyT* keys_out,
    const uint32_t* a0_in, uint32_t* a0_out,
    const uint32_t* a1_in, uint32_t* a1_out,
    const uint64_t* b64_in, uint64_t* b64_out,
    uint32_t n, int byte_idx,
    const uint32_t* bases /* [256] exclusive digit bases */,
    uint32_t* status /* [ntiles*256] */, uint32_t* ticket, uint32_t* error) {
  /* barrier-light ranking (see k_radix_scatter) + decoupled lookback:
   * tiles are ticketed (dispatch order is undefined on CDNA4), per-tile
   * digit totals publish as AGG/INC status words, the lookback overlaps
   * the LDS staging stores. */
  constexpr uint32_t OT = (uint32_t)TILE_ROUNDS * BLK;
  constexpr int OWPB = BLK / WAVE;
  constexpr int RW = OWPB * TILE_ROUNDS;
  constexpr size_t CNT_BYTES = (size_t)RW * RADIX * 2;
  constexpr size_t KEY_BYTES = (size_t)OT * sizeof(KeyT);
  constexpr size_t UNION_BYTES = CNT_BYTES > KEY_BYTES ? CNT_BYTES : KEY_BYTES;
  __shared__ __attribute__((aligned(16))) uint8_t u_raw[UNION_BYTES];
  __shared__ uint32_t ls_a0[OT];
  __shared__ uint32_t ls_a1[HAS_A1 ? OT : 1];
  __shared__ uint64_t ls_b64[HAS_B64 ? OT : 1];
  __shared__ uint32_t tileoff[RADIX];
  __shared__ uint32_t excl[RADIX];
  __shared__ uint32_t s_tile;
  uint16_t* cnt = (uint16_t*)u_raw;
  KeyT* ls_key = (KeyT*)u_raw;
  if (threadIdx.x == 0) s_tile = atomicAdd(ticket, 1u);
  for (uint32_t i = threadIdx.x; i < (uint32_t)RW * RADIX / 2; i += blockDim.x)
    ((uint32_t*)cnt)[i] = 0;
  __syncthreads();
  const uint32_t tile = s_tile;
  const uint32_t start = tile * OT;
  const uint32_t end = min(start + OT, n);
  const uint32_t count = (start < n) ? (end - start) : 0;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wv = threadIdx.x / WAVE;
  const uint64_t lt_mask = (lane == 0) ? 0ull : (~0ull >> (64 - lane));
  KeyT my_key[TILE_ROUNDS];
  uint32_t my_a0[TILE_ROUNDS];
  uint32_t my_a1v[HAS_A1 ? TILE_ROUNDS : 1];
  uint64_t my_b64[HAS_B64 ? TILE_ROUNDS : 1];
  uint16_t my_rank[TILE_ROUNDS];
  uint16_t my_dig[TILE_ROUNDS];
  /* load phase first: every round's gathers issue back-to-back so the
     memory latency is paid once per tile, not once per ranking round */
  #pragma unroll
  for (int r = 0; r < TILE_ROUNDS; r++) {
    uint32_t i = start + (uint32_t)r * BLK + threadIdx.x;
    bool active = i < end;
    my_key[r] = active ? keys_in[i] : (KeyT)0;
    my_a0[r] = active ? a0_in[i] : 0;
    if (HAS_A1) my_a1v[r] = active ? a1_in[i] : 0;
    if (HAS_B64) my_b64[r] = active ? b64_in[i] : 0;
  }
  for (int r = 0; r < TILE_ROUNDS; r++) {
    uint32_t i = start + (uint32_t)r * BLK + threadIdx.x;
    bool active = i < end;
    uint32_t d = active ? ((uint32_t)(my_key[r] >> (8 * byte_idx)) & 0xFF) : 0u;
    uint64_t m = ~0ull;
    for (int b = 0; b < 8; b++) {
      uint64_t bb = __ballot((d >> b) & 1);
      m &= ((d >> b) & 1) ? bb : ~bb;
    }
    uint64_t act = __ballot(active);
    m &= act;
    uint32_t lane_rank = (uint32_t)__popcll(m & lt_mask);
    if (active && lane_rank == 0)
      cnt[(r * OWPB + wv) * RADIX + d] = (uint16_t)__popcll(m);
    my_rank[r] = (uint16_t)lane_rank;
    my_dig[r] = (uint16_t)d;
  }
  __syncthreads();
  /* per-digit exclusive prefix over (round,wave) slots + publish totals */
  if (threadIdx.x < RADIX) {
    uint32_t d = threadIdx.x;
    uint32_t running = 0;
    for (int rw = 0; rw < RW; rw++) {
      uint32_t t = cnt[rw * RADIX + d];
      cnt[rw * RADIX + d] = (uint16_t)running;
      running += t;
    }
    tileoff[d] = running;
    if (tile == 0)
      __hip_atomic_store((os_gu32*)&status[d], (running & OS_CNT_MASK) | OS_INC,
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    else
      __hip_atomic_store((os_gu32*)&status[(uint64_t)tile * RADIX + d],
                         (running & OS_CNT_MASK) | OS_AGG,
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  }
  __syncthreads();
  if (wv == 0) {
    uint32_t v0 = tileoff[4 * lane], v1 = tileoff[4 * lane + 1],
             v2 = tileoff[4 * lane + 2], v3 = tileoff[4 * lane + 3];
    uint32_t sum = v0 + v1 + v2 + v3;
    uint32_t inc = sum;
    for (int s2 = 1; s2 < WAVE; s2 <<= 1) {
      uint32_t t = __shfl_up(inc, s2);
      if (lane >= s2) inc += t;
    }
    uint32_t base0 = inc - sum;
    tileoff[4 * lane] = base0;
    tileoff[4 * lane + 1] = base0 + v0;
    tileoff[4 * lane + 2] = base0 + v0 + v1;
    tileoff[4 * lane + 3] = base0 + v0 + v1 + v2;
  }
  __syncthreads();
  uint16_t my_slot[TILE_ROUNDS];
  for (int r = 0; r < TILE_ROUNDS; r++) {
    uint32_t d = my_dig[r];
    my_slot[r] = (uint16_t)(tileoff[d] + cnt[(r * OWPB + wv) * RADIX + d] +
                            my_rank[r]);
  }
  __syncthreads(); /* cnt consumed; region becomes the key staging */
  for (int r = 0; r < TILE_ROUNDS; r++) {
    uint32_t i = start + (uint32_t)r * BLK + threadIdx.x;
    if (i < end) {
      uint32_t slot = my_slot[r];
      ls_key[slot] = my_key[r];
      ls_a0[slot] = my_a0[r];
      if (HAS_A1) ls_a1[slot] = my_a1v[r];
      if (HAS_B64) ls_b64[slot] = my_b64[r];
    }
  }
  /* lookback overlapped after placement */
  if (threadIdx.x < RADIX) {
    uint32_t d = threadIdx.x;
    uint32_t e = 0;
    if (tile > 0) {
      int64_t p = (int64_t)tile - 1;
      uint32_t spins = 0;
      while (p >= 0) {
        uint32_t v = __hip_atomic_load((os_gu32*)&status[(uint64_t)p * RADIX + d],
                                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        if (v & OS_INC) { e += v & OS_CNT_MASK; break; }
        if (v & OS_AGG) { e += v & OS_CNT_MASK; p--; continue; }
        if (++spins > 100000000u) { atomicAdd(error, 1u); break; }
        __builtin_amdgcn_s_sleep(4);
      }
      uint32_t tot = ((d + 1 < RADIX) ? tileoff[d + 1] : count) - tileoff[d];
      __hip_atomic_store((os_gu32*)&status[(uint64_t)tile * RADIX + d],
                         ((e + tot) & OS_CNT_MASK) | OS_INC,
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    }
    excl[d] = e;
  }
  __syncthreads();
  for (uint32_t j = threadIdx.x; j < count; j += blockDim.x) {
    KeyT k = ls_key[j];
    uint32_t d = (uint32_t)(k >> (8 * byte_idx)) & 0xFF;
    uint32_t pos = bases[d] + excl[d] + (j - tileoff[d]);
    keys_out[pos] = k;
    a0_out[pos] = ls_a0[j];
    if (HAS_A1) a1_out[pos] = ls_a1[j];
    if (HAS_B64) b64_out[pos] = ls_b64[j];
  }
}

/* ---- generic exclusive scan over u64 (sizes -> offsets) ---- */
#define SCAN_ITEMS 8
#define SCAN_TILE (BLOCK * SCAN_ITEMS)
__global__ void k_scan_partials(const uint64_t* in, uint64_t* out,
                                uint64_t* block_sums, uint32_t n) {
  __shared__ uint64_t lds[BLOCK];
  uint32_t base = blockIdx.x * SCAN_TILE;
  uint64_t vals[SCAN_ITEMS];
  uint64_t sum = 0;
  for (int j = 0; j < SCAN_ITEMS; j++) {
    uint32_t i = base + threadIdx.x * SCAN_ITEMS + j;
    vals[j] = (i < n) ? in[i] : 0;
    sum += vals[j];
  }
  lds[threadIdx.x] = sum;
  __syncthreads();
  for (int s = 1; s < BLOCK; s <<= 1) {
    uint64_t t = (threadIdx.x >= (uint32_t)s) ? lds[threadIdx.x - s] : 0;
    __syncthreads();
    lds[threadIdx.x] += t;
    __syncthreads();
  }
  uint64_t excl = lds[threadIdx.x] - sum;
  for (int j = 0; j < SCAN_ITEMS; j++) {
    uint32_t i = base + threadIdx.x * SCAN_ITEMS + j;
    if (i < n) { out[i] = excl; excl += vals[j]; }
  }
  if (threadIdx.x == BLOCK - 1) block_sums[blockIdx.x] = lds[BLOCK - 1];
}
__global__ void k_scan_add(uint64_t* out, const uint64_t* block_offsets, uint32_t n) {
  uint32_t base = blockIdx.x * SCAN_TILE;
  uint64_t add = block_offsets[blockIdx.x];
  for (uint32_t i = base + threadIdx.x; i < min(base + SCAN_TILE, n); i += blockDim.x)
    out[i] += add;
}

/* single-pass u64 exclusive scan with decoupled lookback (ticketed tiles,
 * same protocol as k_onesweep_pass): one read + one write of the data
 * instead of the 3-kernel partials/scan/add chain.  Status word: u64 with
 * flags in the top 2 bits (values here are byte totals < 2^40). */
#define OSS_AGG (1ull << 62)
#define OSS_INC (2ull << 62)
#define OSS_VAL (OSS_AGG - 1)
typedef unsigned long long __attribute__((address_space(1))) os_gu64;
__global__ __launch_bounds__(BLOCK) void k_scan_lookback(
    const uint64_t* in, uint64_t* out, uint32_t n,
    uint64_t* status /* [ntiles] */, uint32_t* ticket, uint32_t* error,
    uint64_t* total_out) {
  __shared__ uint64_t wsum[WPB + 1];
  __shared__ uint32_t s_tile;
  __shared__ uint64_t s_excl;
  if (threadIdx.x == 0) s_tile = atomicAdd(ticket, 1u);
  __syncthreads();
  const uint32_t tile = s_tile;
  const uint32_t base = tile * SCAN_TILE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wv = threadIdx.x / WAVE;
  uint64_t vals[SCAN_ITEMS];
  uint64_t sum = 0;
  #pragma unroll
  for (int j = 0; j < SCAN_ITEMS; j++) {
    uint32_t i = base + threadIdx.x * SCAN_ITEMS + j;
    vals[j] = (i < n) ? in[i] : 0;
    sum += vals[j];
  }
  /* wave-shuffle inclusive scan + tiny cross-wave fixup (the round-1
     Hillis-Steele block scan cost 16 barriers per tile) */
  uint64_t inc = sum;
  for (int s2 = 1; s2 < WAVE; s2 <<= 1) {
    uint64_t t = __shfl_up((unsigned long long)inc, s2);
    if (lane >= s2) inc += t;
  }
  if (lane == WAVE - 1) wsum[wv] = inc;
  __syncthreads();
  if (threadIdx.x == 0) {
    uint64_t run = 0;
    for (int w = 0; w < WPB; w++) { uint64_t t = wsum[w]; wsum[w] = run; run += t; }
    wsum[WPB] = run;
  }
  __syncthreads();
  uint64_t my_incl = wsum[wv] + inc; /* inclusive prefix incl. my sum */
  uint64_t block_total = wsum[WPB];
  if (threadIdx.x == 0) {
    if (tile == 0) {
      __hip_atomic_store((os_gu64*)&status[0], OSS_INC | block_total,
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      s_excl = 0;
    } else {
      __hip_atomic_store((os_gu64*)&status[tile], OSS_AGG | block_total,
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      uint64_t e = 0;
      int64_t p = (int64_t)tile - 1;
      uint32_t spins = 0;
      while (p >= 0) {
        uint64_t v = __hip_atomic_load((os_gu64*)&status[p],
                                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        if (v & OSS_INC) { e += v & OSS_VAL; break; }
        if (v & OSS_AGG) { e += v & OSS_VAL; p--; continue; }
        if (++spins > 100000000u) { atomicAdd(error, 1u); break; }
        __builtin_amdgcn_s_sleep(4);
      }
      __hip_atomic_store((os_gu64*)&status[tile], OSS_INC | (e + block_total),
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      s_excl = e;
    }
  }
  __syncthreads();
  uint64_t excl = s_excl + my_incl - sum;
  #pragma unroll
  for (int j = 0; j < SCAN_ITEMS; j++) {
    uint32_t i = base + threadIdx.x * SCAN_ITEMS + j;
    if (i < n) { out[i] = excl; excl += vals[j]; }
  }
  if (total_out && threadIdx.x == 0 && (uint64_t)base + SCAN_TILE >= n)
    *total_out = s_excl + block_total;
}

/* ---- refinement ---- */
/* eq[i] = 1 if sorted element i has same (still-ambiguous) key prefix as i-1 */
__global__ void k_eq_init2(const uint64_t* skeys, const uint64_t* lo,
                           uint8_t* eq, int pbits, uint32_t* parts,
                           uint32_t n) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x) {
    uint64_t k = skeys[i];
    eq[i] = (i > 0 && k == skeys[i - 1] && lo[i] == lo[i - 1]) ? 1 : 0;
    parts[i] = pbits ? (uint32_t)(k >> (64 - pbits)) : 0;
  }
}
__global__ void k_eq_init(const uint64_t* skeys, uint8_t* eq, int pbits,
                          uint32_t* parts, uint32_t n) {
  /* also materializes the per-position partition ids (composite top bits) —
     skey positions are stable under refinement (intra-run permutations swap
     EQUAL composites), so this single read serves both */
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x) {
    uint64_t k = skeys[i];
    eq[i] = (i > 0 && k == skeys[i - 1]) ? 1 : 0;
    parts[i] = pbits ? (uint32_t)(k >> (64 - pbits)) : 0;
  }
}
/* inrun[i] = eq[i] || eq[i+1]; runstart[i] = inrun && !eq[i].
 * Both 0/1 flags ride ONE u64 as (inrun << 32) | runstart so a single
 * scan_u64 produces both prefix sums — each field's total is <= n <= 2^32-1
 * so the low field can never carry into the high one. */
__global__ void k_run_flags(const uint8_t* eq, uint64_t* packed, uint32_t n) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x) {
    uint64_t in = (eq[i] || (i + 1 < n && eq[i + 1])) ? 1 : 0;
    packed[i] = (in << 32) | (uint64_t)(in && !eq[i]);
  }
}
/* level keys for EVERY record in original-record order (coalesced off/klen/
 * data reads): used when most elements are still ambiguous — a random
 * per-element gather through sidx is latency-bound, the dense build + one
 * 8B permuted read is not */
__global__ void k_build_lkeys(RecTable rt, int level_byte0, int use_len_level,
                              int ser_mode, uint64_t* lk0, uint32_t n) {
  for (uint32_t g = blockIdx.x * blockDim.x + threadIdx.x; g < n;
       g += gridDim.x * blockDim.x) {
    RecView v = rt_view(rt, g);
    const uint8_t* src = ser_mode ? v.key : v.content;
    uint32_t slen = ser_mode ? v.klen : v.clen;
    uint64_t k = 0;
    if (use_len_level) {
      k = slen;
    } else {
      for (int b = 0; b < 8; b++) {
        uint32_t cb = (uint32_t)(level_byte0 + b);
        uint8_t byte = (cb < slen) ? src[cb] : 0;
        k |= (uint64_t)byte << (56 - 8 * b);
      }
    }
    lk0[g] = k;
  }
}
/* compact ambiguous elements; seg = run rank; gather level key (from lk0
 * when prebuilt, else straight from the record table) */
__global__ void k_compact_refine(RecTable rt, const uint32_t* sidx,
                                 const uint8_t* eq,
                                 const uint64_t* packed_scan,
                                 uint32_t n, int level_byte0, int use_len_level,
                                 int ser_mode, const uint64_t* lk0,
                                 uint64_t* lkey, uint32_t* seg, uint32_t* pos) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x) {
    uint8_t in = eq[i] || (i + 1 < n && eq[i + 1]);
    if (!in) continue;
    uint64_t ps = packed_scan[i];
    uint32_t j = (uint32_t)(ps >> 32);
    /* run rank via the EXCLUSIVE runstart scan: rank = scan + is_start - 1 */
    uint32_t is_start = (eq[i] == 0) ? 1u : 0u;
    uint32_t sg = (uint32_t)(ps & 0xFFFFFFFFu) + is_start - 1;
    uint64_t k = 0;
    if (lk0) {
      k = lk0[sidx[i]];
    } else if (use_len_level) {
      RecView v = rt_view(rt, sidx[i]);
      k = ser_mode ? v.klen : v.clen;
    } else {
      RecView v = rt_view(rt, sidx[i]);
      const uint8_t* src = ser_mode ? v.key : v.content;
      uint32_t slen = ser_mode ? v.klen : v.clen;
      for (int b = 0; b < 8; b++) {
        uint32_t cb = (uint32_t)(level_byte0 + b);
        uint8_t byte = (cb < slen) ? src[cb] : 0;
        k |= (uint64_t)byte << (56 - 8 * b);
      }
    }
    lkey[j] = k;
    seg[j] = sg;
    pos[j] = i;
  }
}
/* fused run-flags + decoupled-lookback scan + compaction: one kernel
 * replaces k_run_flags + scan_u64 + k_compact_refine (the separate chain
 * wrote and re-read an 8n flag array and re-read eq — ~24 B/element of
 * scratch traffic; C3's refinement was the dominant profile block).
 * Packed u64 counters: (inrun << 32) | runstart, same protocol as
 * k_scan_lookback. */
__global__ __launch_bounds__(BLOCK) void k_refine_compact_lb(
    RecTable rt, const uint32_t* sidx, const uint8_t* eq, uint32_t n,
    int level_byte0, int use_len_level, int ser_mode, const uint64_t* lk0,
    uint64_t* lkey, uint32_t* seg, uint32_t* pos,
    uint64_t* status, uint32_t* ticket, uint32_t* error, uint64_t* total_out) {
  __shared__ uint64_t wsum[WPB + 1];
  __shared__ uint32_t s_tile;
  __shared__ uint64_t s_excl;
  if (threadIdx.x == 0) s_tile = atomicAdd(ticket, 1u);
  __syncthreads();
  const uint32_t tile = s_tile;
  const uint32_t base = tile * SCAN_TILE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wv = threadIdx.x / WAVE;
  uint8_t my_eq[SCAN_ITEMS];
  uint8_t my_in[SCAN_ITEMS];
  uint64_t vals[SCAN_ITEMS];
  uint64_t sum = 0;
  {
    /* eq flags for the thread's 8 elements ride ONE aligned u64 load plus
       one byte (the i+1 lookahead) — 16 serial byte loads left the kernel
       93% wave-parked (r02 PMC) */
    uint32_t i0b = base + threadIdx.x * SCAN_ITEMS;
    uint64_t w = 0;
    uint8_t nxt = 0;
    if (i0b + SCAN_ITEMS <= n) {
      w = *(const uint64_t*)(eq + i0b);
      nxt = (i0b + SCAN_ITEMS < n) ? eq[i0b + SCAN_ITEMS] : 0;
    } else if (i0b < n) {
      for (uint32_t j = 0; i0b + j < n && j < SCAN_ITEMS; j++)
        w |= (uint64_t)eq[i0b + j] << (8 * j);
    }
    #pragma unroll
    for (int j = 0; j < SCAN_ITEMS; j++) {
      uint32_t i = i0b + j;
      uint8_t e = (uint8_t)(w >> (8 * j));
      uint8_t nx = (j + 1 < SCAN_ITEMS) ? (uint8_t)(w >> (8 * (j + 1))) : nxt;
      uint8_t in = (i < n) && (e || ((i + 1 < n) && nx));
      my_eq[j] = e;
      my_in[j] = in;
      uint64_t rs = (in && !e) ? 1 : 0;
      vals[j] = ((uint64_t)in << 32) | rs;
      sum += vals[j];
    }
  }
  uint64_t inc = sum;
  for (int s2 = 1; s2 < WAVE; s2 <<= 1) {
    uint64_t t = __shfl_up((unsigned long long)inc, s2);
    if (lane >= s2) inc += t;
  }
  if (lane == WAVE - 1) wsum[wv] = inc;
  __syncthreads();
  if (threadIdx.x == 0) {
    uint64_t run = 0;
    for (int w = 0; w < WPB; w++) { uint64_t t = wsum[w]; wsum[w] = run; run += t; }
    wsum[WPB] = run;
  }
  __syncthreads();
  uint64_t my_incl = wsum[wv] + inc;
  uint64_t block_total = wsum[WPB];
  if (threadIdx.x == 0) {
    if (tile == 0) {
      __hip_atomic_store((os_gu64*)&status[0], OSS_INC | block_total,
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      s_excl = 0;
    } else {
      __hip_atomic_store((os_gu64*)&status[tile], OSS_AGG | block_total,
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      uint64_t e = 0;
      int64_t p = (int64_t)tile - 1;
      uint32_t spins = 0;
      while (p >= 0) {
        uint64_t v = __hip_atomic_load((os_gu64*)&status[p],
                                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        if (v & OSS_INC) { e += v & OSS_VAL; break; }
        if (v & OSS_AGG) { e += v & OSS_VAL; p--; continue; }
        if (++spins > 100000000u) { atomicAdd(error, 1u); break; }
        __builtin_amdgcn_s_sleep(4);
      }
      __hip_atomic_store((os_gu64*)&status[tile], OSS_INC | (e + block_total),
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      s_excl = e;
    }
  }
  __syncthreads();
  uint64_t excl = s_excl + my_incl - sum;
  /* prefetch the dense-key gathers for all 8 elements up front (the
     guarded in-loop loads serialized one gather at a time) */
  uint64_t pre_k[SCAN_ITEMS];
  if (lk0) {
    uint32_t pre_ix[SCAN_ITEMS];
    #pragma unroll
    for (int j = 0; j < SCAN_ITEMS; j++) {
      uint32_t i = base + threadIdx.x * SCAN_ITEMS + j;
      pre_ix[j] = sidx[i < n ? i : n - 1];
    }
    #pragma unroll
    for (int j = 0; j < SCAN_ITEMS; j++) pre_k[j] = lk0[pre_ix[j]];
  }
  #pragma unroll
  for (int j = 0; j < SCAN_ITEMS; j++) {
    uint32_t i = base + threadIdx.x * SCAN_ITEMS + j;
    if (i < n && my_in[j]) {
      uint32_t cj = (uint32_t)(excl >> 32);
      uint32_t is_start = my_eq[j] ? 0u : 1u;
      uint32_t sg = (uint32_t)(excl & 0xFFFFFFFFu) + is_start - 1;
      uint64_t k;
      if (lk0) {
        k = pre_k[j];
      } else if (use_len_level) {
        RecView v = rt_view(rt, sidx[i]);
        k = ser_mode ? v.klen : v.clen;
      } else {
        RecView v = rt_view(rt, sidx[i]);
        const uint8_t* src = ser_mode ? v.key : v.content;
        uint32_t slen = ser_mode ? v.klen : v.clen;
        k = 0;
        for (int b = 0; b < 8; b++) {
          uint32_t cb = (uint32_t)(level_byte0 + b);
          uint8_t byte = (cb < slen) ? src[cb] : 0;
          k |= (uint64_t)byte << (56 - 8 * b);
        }
      }
      lkey[cj] = k;
      seg[cj] = sg;
      pos[cj] = i;
    }
    excl += vals[j];
  }
  if (total_out && threadIdx.x == 0 && (uint64_t)base + SCAN_TILE >= n)
    *total_out = s_excl + block_total;
}

/* scatter refined order back: new_sidx[pos_sorted[j]] stays — we permute the
 * POSITIONS: output[orig_pos_slot_j] where slot order = sorted compact order.
 * We write: for compact rank j (after sort), its element moves to the j-th
 * in-run position.  Since runs are contiguous and seg-major sorted order keeps
 * run grouping, the j-th compact slot in sorted order corresponds to the j-th
 * in-run position overall (pos_sorted ascending within each run, runs in
 * order).  So: new_sidx[ pos_of_slot[j] ] = old_sidx[ pos[j] ] where
 * pos_of_slot = the ORIGINAL pos array in its pre-sort (position-ascending)
 * order.  We pass both. */
/* two-phase in-place variant: {pos} and {slotpos} are the same position SET
 * (the in-run slots), so gathering old values first then scattering touches
 * only m slots — no full-n index copy */
__global__ void k_refine_gather(const uint32_t* pos_sorted, const uint32_t* sidx,
                                uint32_t* tmp, uint32_t m) {
  for (uint32_t j = blockIdx.x * blockDim.x + threadIdx.x; j < m;
       j += gridDim.x * blockDim.x)
    tmp[j] = sidx[pos_sorted[j]];
}
__global__ void k_refine_scatter(const uint32_t* slot_positions, const uint32_t* tmp,
                                 uint32_t* sidx, uint32_t m) {
  for (uint32_t j = blockIdx.x * blockDim.x + threadIdx.x; j < m;
       j += gridDim.x * blockDim.x)
    sidx[slot_positions[j]] = tmp[j];
}
__global__ void k_refine_apply(const uint32_t* pos_sorted_elements /* pos[j] after sort */,
                               const uint32_t* slot_positions /* ascending positions */,
                               const uint32_t* old_sidx, uint32_t* new_sidx,
                               uint32_t m) {
  for (uint32_t j = blockIdx.x * blockDim.x + threadIdx.x; j < m;
       j += gridDim.x * blockDim.x)
    new_sidx[slot_positions[j]] = old_sidx[pos_sorted_elements[j]];
}
/* update eq inside runs after a refine level; counts survivors so the next
 * level can skip its full-n ambiguity pass */
__global__ void k_eq_update(const uint64_t* lkey_sorted, const uint32_t* seg_sorted,
                            const uint32_t* slot_positions, uint8_t* eq, uint32_t m,
                            uint32_t* survivors) {
  __shared__ uint32_t s;
  if (threadIdx.x == 0) s = 0;
  __syncthreads();
  uint32_t loc = 0;
  for (uint32_t j = blockIdx.x * blockDim.x + threadIdx.x; j < m;
       j += gridDim.x * blockDim.x) {
    uint8_t e = 0;
    if (j > 0 && seg_sorted[j] == seg_sorted[j - 1] && lkey_sorted[j] == lkey_sorted[j - 1])
      e = 1;
    /* an element that was a run start keeps eq=0; others get refined eq */
    eq[slot_positions[j]] = e;
    loc += e;
  }
  atomicAdd(&s, loc);
  __syncthreads();
  if (threadIdx.x == 0 && s) atomicAdd(survivors, s);
}
__global__ void k_count_nonzero_u8(const uint8_t* a, uint32_t n, uint32_t* out) {
  __shared__ uint32_t s;
  if (threadIdx.x == 0) s = 0;
  __syncthreads();
  uint32_t loc = 0;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x)
    loc += a[i];
  atomicAdd(&s, loc);
  __syncthreads();
  if (threadIdx.x == 0) atomicAdd(out, s);
}

/* ---- combiner (SUM_INT) ----
 * Fold runs of equal keys (eq flags) into one record with the 4-byte
 * big-endian IntWritable sum (java int wrap) — runCombineProcessor
 * restated (PipelinedSorter.java:602-609,816-821). */
__global__ void k_combine_mark(const uint8_t* eq, uint64_t* runstart, uint32_t n) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x)
    runstart[i] = eq[i] ? 0 : 1;
}
__global__ void k_combine_pos(const uint8_t* eq, const uint64_t* rs_scan,
                              uint32_t* pos, uint64_t* lens2, RecTable rt,
                              const uint32_t* sidx, uint32_t n) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x) {
    if (!eq[i]) {
      uint32_t j = (uint32_t)rs_scan[i];
      pos[j] = i;
      RecView v = rt_view(rt, sidx[i]);
      lens2[j] = (uint64_t)v.klen + 4;
    }
  }
}
__global__ void k_combine_fold(RecTable rt, const uint32_t* sidx, const uint32_t* pos,
                               const uint64_t* off2, const uint32_t* parts_in,
                               uint8_t* data2, uint32_t* klen2, uint32_t* parts2,
                               uint32_t M, uint32_t n) {
  uint32_t wave = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  uint32_t lane = threadIdx.x & (WAVE - 1);
  uint32_t nwaves = (gridDim.x * blockDim.x) / WAVE;
  for (uint32_t j = wave; j < M; j += nwaves) {
    uint32_t i0 = pos[j];
    uint32_t i1 = (j + 1 < M) ? pos[j + 1] : n;
    RecView k0 = rt_view(rt, sidx[i0]);
    /* lane-parallel partial sums over the run */
    uint32_t sum = 0;
    for (uint32_t i = i0 + lane; i < i1; i += WAVE) {
      RecView v = rt_view(rt, sidx[i]);
      sum += ((uint32_t)v.val[0] << 24) | ((uint32_t)v.val[1] << 16) |
             ((uint32_t)v.val[2] << 8) | (uint32_t)v.val[3];
    }
    for (int sh = 32; sh >= 1; sh >>= 1) sum += __shfl_xor(sum, sh);
    uint8_t* w = data2 + off2[j];
    for (uint32_t b = lane; b < k0.klen; b += WAVE) w[b] = k0.key[b];
    if (lane == 0) {
      w[k0.klen + 0] = (uint8_t)(sum >> 24);
      w[k0.klen + 1] = (uint8_t)(sum >> 16);
      w[k0.klen + 2] = (uint8_t)(sum >> 8);
      w[k0.klen + 3] = (uint8_t)sum;
      klen2[j] = k0.klen;
      parts2[j] = parts_in[i0];
    }
  }
}
__global__ void k_iota(uint32_t* a, uint32_t n) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x)
    a[i] = i;
}

/* ---- merge path (stable pairwise merge of sorted composite streams) ----
 * Replaces the union re-sort of already-sorted spill segments
 * (VERDICT r1 missing #1; TezMerger.java:466-706 semantics: the k-way
 * merge of per-spill sorted runs, ties to the lower segment).  Each spill
 * retains its sorted (composite, original-id) arrays; flush merges them in
 * ceil(log2 k) passes of 24 B/element instead of ~8 radix passes over the
 * union.  Masks widen per-stream composites to the common coarse mask
 * (adaptive per-spill sort widths — DESIGN.md §4); `add` rebases per-spill
 * record ids to global ids.  Stability: A (lower segment) wins ties, so the
 * merged run order for equal composites is (segment, in-segment position) =
 * global-id order — exactly the stable union re-sort's order, which keeps
 * the refinement stage's output bit-identical. */
#define MP_IPT 16
#define MP_BLOCK 256
#define MP_TILE (MP_IPT * MP_BLOCK)

/* diagonal split: # taken from A at output rank D (A wins ties) */
__device__ __forceinline__ uint32_t d_mp_diag(
    const uint64_t* ka, uint64_t maskA, uint32_t na,
    const uint64_t* kb, uint64_t maskB, uint32_t nb, uint64_t D) {
  uint32_t lo = (D > nb) ? (uint32_t)(D - nb) : 0;
  uint32_t hi = (D < na) ? (uint32_t)D : na;
  while (lo < hi) {
    uint32_t mid = lo + ((hi - lo) >> 1);
    if ((ka[mid] & maskA) <= (kb[D - 1 - mid] & maskB)) lo = mid + 1;
    else hi = mid;
  }
  return lo;
}

/* nondecreasing check over composites (validates add_sorted_segment input) */
__global__ void k_check_sorted(const uint64_t* k, uint32_t n, uint32_t* bad) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x)
    if (i > 0 && k[i] < k[i - 1]) atomicAdd(bad, 1u);
}

/* gather src[idx[i]] into dst[i], keeping dst's top (partition) bits:
 * used to rebuild a spill's retained composites in serialized form without
 * recomputing its partition placement (explicit partitioners), and — with
 * himask 0 — as a plain permutation gather (lo-key retention). */
__global__ void k_gather_merge_hi(const uint64_t* src, const uint32_t* idx,
                                  uint64_t* dst, uint64_t himask, uint32_t n) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x)
    dst[i] = (dst[i] & himask) | (src[idx[i]] & ~himask);
}

/* leaf/pass-through materialization: apply mask + id rebase */
__global__ void k_apply_leaf2(const uint64_t* k, const uint64_t* lo,
                              const uint32_t* p, uint32_t n,
                              uint64_t mask, uint32_t add,
                              uint64_t* ko, uint64_t* loo, uint32_t* po) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x) {
    ko[i] = k[i] & mask;
    loo[i] = lo[i];
    po[i] = p[i] + add;
  }
}

/* 128-bit merge path: key = (hi & mask, lo).  Each spill retains lo = the
 * next 8 comparator-source bytes past the composite (zero-padded), so the
 * merged order resolves ~15 leading key bytes and the flush refinement only
 * sees genuinely long shared prefixes (C3's Zipf words were re-refined over
 * the whole union otherwise — the round-2 profile's dominant block). */
#define MP2_IPT 8
#define MP2_TILE (MP2_IPT * MP_BLOCK) /* 2048x20B; the 4096 tile measured
  2x slower (148 vs 77 ms at C3 1e9: occupancy 3 -> 2 blocks/CU and the
  16-deep serial sub-merge) */
__device__ __forceinline__ uint32_t d_mp2_diag(
    const uint64_t* ka, const uint64_t* la, uint64_t maskA, uint32_t na,
    const uint64_t* kb, const uint64_t* lb, uint64_t maskB, uint32_t nb,
    uint64_t D) {
  uint32_t lo_ = (D > nb) ? (uint32_t)(D - nb) : 0;
  uint32_t hi_ = (D < na) ? (uint32_t)D : na;
  while (lo_ < hi_) {
    uint32_t mid = lo_ + ((hi_ - lo_) >> 1);
    uint64_t ah = ka[mid] & maskA, bh = kb[D - 1 - mid] & maskB;
    bool le = (ah < bh) || (ah == bh && la[mid] <= lb[D - 1 - mid]);
    if (le) lo_ = mid + 1;
    else hi_ = mid;
  }
  return lo_;
}
__global__ void k_mp2_partition(const uint64_t* ka, const uint64_t* la,
                                uint32_t na, uint64_t maskA,
                                const uint64_t* kb, const uint64_t* lb,
                                uint32_t nb, uint64_t maskB,
                                uint32_t nblk, uint32_t tile, uint32_t* splits) {
  uint64_t total = (uint64_t)na + nb;
  for (uint32_t t = blockIdx.x * blockDim.x + threadIdx.x; t <= nblk;
       t += gridDim.x * blockDim.x) {
    uint64_t D = min((uint64_t)t * tile, total);
    splits[t] = d_mp2_diag(ka, la, maskA, na, kb, lb, maskB, nb, D);
  }
}
__global__ __launch_bounds__(MP_BLOCK) void k_merge_path2(
    const uint64_t* ka, const uint64_t* la, const uint32_t* pa, uint32_t na,
    uint64_t maskA, uint32_t addA,
    const uint64_t* kb, const uint64_t* lb, const uint32_t* pb, uint32_t nb,
    uint64_t maskB, uint32_t addB,
    uint64_t* kout, uint64_t* loout, uint32_t* pout,
    const uint32_t* splits /* [nblk+1] from k_mp2_partition */) {
  __shared__ uint64_t ls_k[MP2_TILE];
  __shared__ uint64_t ls_l[MP2_TILE];
  __shared__ uint32_t ls_p[MP2_TILE];
  uint64_t total = (uint64_t)na + nb;
  uint64_t D0 = (uint64_t)blockIdx.x * MP2_TILE;
  if (D0 >= total) return;
  uint64_t D1 = min(D0 + (uint64_t)MP2_TILE, total);
  uint32_t a0 = splits[blockIdx.x], a1 = splits[blockIdx.x + 1];
  uint32_t b0 = (uint32_t)(D0 - a0), b1 = (uint32_t)(D1 - a1);
  uint32_t nA = a1 - a0, nB = b1 - b0;
  for (uint32_t i = threadIdx.x; i < nA; i += blockDim.x) {
    ls_k[i] = ka[a0 + i] & maskA;
    ls_l[i] = la[a0 + i];
    ls_p[i] = pa[a0 + i] + addA;
  }
  for (uint32_t i = threadIdx.x; i < nB; i += blockDim.x) {
    ls_k[nA + i] = kb[b0 + i] & maskB;
    ls_l[nA + i] = lb[b0 + i];
    ls_p[nA + i] = pb[b0 + i] + addB;
  }
  __syncthreads();
  uint32_t r = threadIdx.x * MP2_IPT;
  uint32_t tile_n = (uint32_t)(D1 - D0);
  if (r >= tile_n) return;
  uint32_t cnt = min(tile_n - r, (uint32_t)MP2_IPT);
  uint32_t lo_ = (r > nB) ? r - nB : 0, hi_ = min(r, nA);
  while (lo_ < hi_) {
    uint32_t mid = lo_ + ((hi_ - lo_) >> 1);
    bool le = (ls_k[mid] < ls_k[nA + r - 1 - mid]) ||
              (ls_k[mid] == ls_k[nA + r - 1 - mid] &&
               ls_l[mid] <= ls_l[nA + r - 1 - mid]);
    if (le) lo_ = mid + 1;
    else hi_ = mid;
  }
  uint32_t i = lo_, j = r - lo_;
  uint64_t ok[MP2_IPT], ol[MP2_IPT];
  uint32_t op[MP2_IPT];
  #pragma unroll
  for (uint32_t t = 0; t < MP2_IPT; t++) {
    if (t < cnt) {
      bool takeA = (j >= nB) ||
                   (i < nA && (ls_k[i] < ls_k[nA + j] ||
                               (ls_k[i] == ls_k[nA + j] &&
                                ls_l[i] <= ls_l[nA + j])));
      uint32_t src = takeA ? i++ : nA + (j++);
      ok[t] = ls_k[src];
      ol[t] = ls_l[src];
      op[t] = ls_p[src];
    }
  }
  #pragma unroll
  for (uint32_t t = 0; t < MP2_IPT; t++)
    if (t < cnt) {
      kout[D0 + r + t] = ok[t];
      loout[D0 + r + t] = ol[t];
      pout[D0 + r + t] = op[t];
    }
}

/* ---- emit ---- */
/* Sorted record descriptors: one gather pass per sort; all later emit-side
 * kernels read these coalesced instead of re-gathering off/klen per record
 * (PMC showed 13+ GB of over-fetch per re-gather at n=1e8). */
struct RecDesc {
  uint64_t src;   /* absolute device address of the serialized record */
  uint32_t klen;
  uint32_t vlen;
};
/* IFile body bytes of sorted record i (a run never crosses a partition
 * segment: the previous record's RLE state is invisible to this partition's
 * stream) — fused into the descriptor builders below */
__device__ __forceinline__ uint64_t d_rec_emit_size(
    const RecDesc& v, const uint8_t* same, const uint32_t* parts, uint32_t i) {
  uint8_t prev_same = (i > 0 && parts[i] == parts[i - 1]) ? same[i - 1] : 0;
  if (same[i])
    return (prev_same ? 0 : 1) /* RLE marker */ + d_vint_size(v.vlen) + v.vlen;
  return (prev_same ? 1 : 0) /* V_END */ + d_vint_size(v.klen) + d_vint_size(v.vlen)
         + v.klen + v.vlen;
}
__global__ void k_build_desc(RecTable rt, const uint32_t* sidx /* NULL: identity */,
                             RecDesc* desc, const uint8_t* same,
                             const uint32_t* parts, uint64_t* sizes, uint32_t n) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x) {
    RecView v = rt_view(rt, sidx ? sidx[i] : i);
    RecDesc d;
    d.src = (uint64_t)(uintptr_t)v.key;
    d.klen = v.klen;
    d.vlen = v.vlen;
    desc[i] = d;
    if (sizes) sizes[i] = d_rec_emit_size(d, same, parts, i);
  }
}
/* permute descriptors into sorted order: one 16B random read per record vs
 * k_build_desc's 3-5 scattered off/klen/data reads through sidx (2x less
 * line traffic for variable-length tables) */
__global__ void k_permute_desc(const RecDesc* src, const uint32_t* sidx,
                               RecDesc* dst, const uint8_t* same,
                               const uint32_t* parts, uint64_t* sizes, uint32_t n) {
  /* 2 records per iteration: two independent sidx->desc gather chains */
  uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += 2 * stride) {
    uint32_t j = i + stride;
    uint32_t si = sidx[i];
    uint32_t sj = (j < n) ? sidx[j] : si;
    RecDesc d0 = src[si];
    RecDesc d1 = src[sj];
    dst[i] = d0;
    sizes[i] = d_rec_emit_size(d0, same, parts, i);
    if (j < n) {
      dst[j] = d1;
      sizes[j] = d_rec_emit_size(d1, same, parts, j);
    }
  }
}
/* same-as-prev full-key flags are exactly the final eq[] array.
 * writer-sameness (what the IFile stream encodes as RLE) additionally
 * depends on the rle flag and merge provenance:
 *   same_writer[i] = eq[i] && (writer_rle || cross_spill || spill_was_rle) */
__global__ void k_writer_same(RecTable rt, const uint32_t* sidx, const uint8_t* eq,
                              int writer_rle, const uint8_t* spill_rle /* [nspills] */,
                              uint8_t* same, uint32_t n) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x) {
    uint8_t s = 0;
    if (eq[i]) {
      if (writer_rle) s = 1;
      else {
        int sa = rt_spill_of(rt, sidx[i]);
        int sb = rt_spill_of(rt, sidx[i - 1]);
        if (sa != sb || spill_rle[sa]) s = 1;
      }
    }
    same[i] = s;
  }
}

/* per-record IFile emit (replaces the IFile.Writer.append loop,
 * IFile.java:444-615): batch of 64 records per wave iteration — each lane
 * fetches ONE record's descriptor in parallel, then two records are kept in
 * flight per wave (independent 32-lane halves + a 2-deep rotate pipeline)
 * so gather latency overlaps stores.  Records longer than 128 B take the
 * simple per-record loop. */
__global__ void k_emit_records(const RecDesc* desc, const uint8_t* same,
                                  const uint64_t* scan, const uint32_t* parts,
                                  const uint64_t* seg_payload_start,
                                  const uint64_t* part_scan_base,
                                  uint8_t* out, uint32_t n, int force_simple) {
  constexpr int HB = 4; /* bytes per lane per record (records to 128 B) */
  uint32_t wave = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  uint32_t lane = threadIdx.x & (WAVE - 1);
  uint32_t half = lane >> 5;        /* 0 or 1 */
  uint32_t hl = lane & 31;          /* lane within half */
  uint32_t nwaves = (gridDim.x * blockDim.x) / WAVE;
  for (uint64_t base = (uint64_t)wave * WAVE; base < n;
       base += (uint64_t)nwaves * WAVE) {
    uint32_t i = (uint32_t)base + lane;
    uint64_t my_src = 0, my_dst = 0, my_h0 = 0, my_h1 = 0;
    uint32_t my_len = 0, my_hdr = 0;
    if (i < n) {
      RecDesc v = desc[i];
      uint32_t p = parts[i];
      my_dst = seg_payload_start[p] + (scan[i] - part_scan_base[p]);
      uint8_t prev_same = (i > 0 && parts[i] == parts[i - 1]) ? same[i - 1] : 0;
      uint8_t hdrbuf[16] = {0};
      uint32_t hdr = 0;
      if (same[i]) {
        if (!prev_same) hdrbuf[hdr++] = 0xFE;
        hdr += d_vint_write(hdrbuf + hdr, v.vlen);
        my_src = v.src + v.klen;
        my_len = v.vlen;
      } else {
        if (prev_same) hdrbuf[hdr++] = 0xFD;
        hdr += d_vint_write(hdrbuf + hdr, v.klen);
        hdr += d_vint_write(hdrbuf + hdr, v.vlen);
        my_src = v.src;
        my_len = v.klen + v.vlen;
      }
      my_hdr = hdr;
      for (int b = 0; b < 8; b++) my_h0 |= (uint64_t)hdrbuf[b] << (8 * b);
      for (int b = 8; b < 12; b++) my_h1 |= (uint64_t)hdrbuf[b] << (8 * (b - 8));
    }
    uint32_t nvalid = (n - base < WAVE) ? (uint32_t)(n - base) : WAVE;
    uint32_t maxlen = my_len;
    for (int sh = 32; sh >= 1; sh >>= 1) {
      uint32_t v2 = __shfl_xor(maxlen, sh);
      if (v2 > maxlen) maxlen = v2;
    }
    uint64_t hdr_ok = __ballot(my_hdr <= 8);
    if (nvalid == WAVE && maxlen <= 32 * HB && hdr_ok == ~0ull &&
        !force_simple) {
      /* half h handles records 2r+h; 2-deep rotate pipeline per half.
       * Byte-granular moves are the measured optimum here: a word-funnel
       * STORE variant ran 17.7 vs 10.3 ms (register pressure), and a
       * one-window-load + shuffle-redistribution GATHER variant was
       * neutral (10.6 ms) — L1 already serves the byte re-reads.
       * (dst | hdr | len) ride ONE packed word and headers <= 8 B ride h0
       * alone: 3 shuffles per rotate instead of 6. */
      uint64_t my_meta = (my_dst << 12) | ((uint64_t)my_hdr << 8) | my_len;
      uint32_t r0 = half;           /* first record index for this half */
      uint64_t src0 = __shfl(my_src, r0);
      uint64_t h00 = __shfl(my_h0, r0);
      uint64_t meta0 = __shfl(my_meta, r0);
      uint8_t b0[HB];
      {
        uint32_t len0 = (uint32_t)(meta0 & 0xFF);
        const uint8_t* sp = (const uint8_t*)(uintptr_t)src0;
#pragma unroll
        for (int k = 0; k < HB; k++)
          b0[k] = (hl + k * 32 < len0) ? sp[hl + k * 32] : 0;
      }
      for (uint32_t r = 0; r < 32; r++) {
        uint32_t rn = 2 * (r + 1) + half;
        uint64_t src1 = 0, h01 = 0, meta1 = 0;
        uint8_t b1[HB] = {0};
        if (rn < WAVE) {
          src1 = __shfl(my_src, rn);
          h01 = __shfl(my_h0, rn);
          meta1 = __shfl(my_meta, rn);
          uint32_t len1 = (uint32_t)(meta1 & 0xFF);
          const uint8_t* sp = (const uint8_t*)(uintptr_t)src1;
#pragma unroll
          for (int k = 0; k < HB; k++)
            b1[k] = (hl + k * 32 < len1) ? sp[hl + k * 32] : 0;
        }
        uint32_t len0 = (uint32_t)(meta0 & 0xFF);
        uint32_t hdr0 = (uint32_t)((meta0 >> 8) & 0xF);
        uint8_t* w = out + (meta0 >> 12);
        if (hl < hdr0) w[hl] = (uint8_t)(h00 >> (8 * hl));
        w += hdr0;
#pragma unroll
        for (int k = 0; k < HB; k++)
          if (hl + k * 32 < len0) w[hl + k * 32] = b0[k];
        src0 = src1; h00 = h01; meta0 = meta1;
#pragma unroll
        for (int k = 0; k < HB; k++) b0[k] = b1[k];
      }
    } else {
      for (uint32_t r = 0; r < nvalid; r++) {
        uint64_t src = __shfl(my_src, r);
        uint64_t dsto = __shfl(my_dst, r);
        uint64_t h0 = __shfl(my_h0, r);
        uint64_t h1 = __shfl(my_h1, r);
        uint32_t len = __shfl(my_len, r);
        uint32_t hdr = __shfl(my_hdr, r);
        uint8_t* w = out + dsto;
        if (lane < hdr)
          w[lane] = (lane < 8) ? (uint8_t)(h0 >> (8 * lane))
                               : (uint8_t)(h1 >> (8 * (lane - 8)));
        w += hdr;
        const uint8_t* sp = (const uint8_t*)(uintptr_t)src;
        for (uint32_t b = lane; b < len; b += WAVE) w[b] = sp[b];
      }
    }
  }
}

/* Staged-gather emit v2: each wave takes 64 records; every lane issues ALL
 * of its own record's dword loads at once (64 records x ~23 loads of MLP —
 * the rotate pipeline kept only 2 records in flight and sat 70% parked,
 * PMC r2), staging the raw source words into LDS; the drain then runs the
 * proven byte-store path against LDS instead of global.  Records longer
 * than EMIT2_CAP bytes fall back to the per-record global loop. */
#define EMIT2_CAP 128
#define EMIT2_W (EMIT2_CAP / 4 + 2)
__global__ __launch_bounds__(BLOCK) void k_emit_records_v2(
    const RecDesc* desc, const uint8_t* same, const uint64_t* scan,
    const uint32_t* parts, const uint64_t* seg_payload_start,
    const uint64_t* part_scan_base, uint8_t* out, uint32_t n) {
  __shared__ uint32_t stage[WPB][WAVE][EMIT2_W];
  uint32_t wave = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  uint32_t lane = threadIdx.x & (WAVE - 1);
  uint32_t wv = threadIdx.x / WAVE;
  uint32_t nwaves = (gridDim.x * blockDim.x) / WAVE;
  for (uint64_t base = (uint64_t)wave * WAVE; base < n;
       base += (uint64_t)nwaves * WAVE) {
    uint32_t i = (uint32_t)base + lane;
    uint64_t my_src = 0, my_dst = 0, my_h0 = 0, my_h1 = 0;
    uint32_t my_len = 0, my_hdr = 0, my_ph = 0;
    if (i < n) {
      RecDesc v = desc[i];
      uint32_t p = parts[i];
      my_dst = seg_payload_start[p] + (scan[i] - part_scan_base[p]);
      uint8_t prev_same = (i > 0 && parts[i] == parts[i - 1]) ? same[i - 1] : 0;
      uint8_t hdrbuf[16] = {0};
      uint32_t hdr = 0;
      if (same[i]) {
        if (!prev_same) hdrbuf[hdr++] = 0xFE;
        hdr += d_vint_write(hdrbuf + hdr, v.vlen);
        my_src = v.src + v.klen;
        my_len = v.vlen;
      } else {
        if (prev_same) hdrbuf[hdr++] = 0xFD;
        hdr += d_vint_write(hdrbuf + hdr, v.klen);
        hdr += d_vint_write(hdrbuf + hdr, v.vlen);
        my_src = v.src;
        my_len = v.klen + v.vlen;
      }
      my_hdr = hdr;
      for (int b = 0; b < 8; b++) my_h0 |= (uint64_t)hdrbuf[b] << (8 * b);
      for (int b = 8; b < 12; b++) my_h1 |= (uint64_t)hdrbuf[b] << (8 * (b - 8));
    }
    uint32_t nvalid = (n - base < WAVE) ? (uint32_t)(n - base) : WAVE;
    uint32_t maxlen = my_len;
    for (int sh = 32; sh >= 1; sh >>= 1) {
      uint32_t v2 = __shfl_xor(maxlen, sh);
      if (v2 > maxlen) maxlen = v2;
    }
    if (maxlen <= EMIT2_CAP) {
      /* stage: aligned dword loads from src&~3 — ALL records in flight */
      my_ph = (uint32_t)(my_src & 3);
      const uint32_t* sw = (const uint32_t*)(my_src & ~3ull);
      uint32_t nw = (my_ph + my_len + 3) >> 2;
      uint32_t* row = stage[wv][lane];
      #pragma unroll 4
      for (uint32_t k = 0; k < nw; k++) row[k] = sw[k];
      /* drain: per record, whole wave cooperates; source = LDS */
      for (uint32_t r = 0; r < nvalid; r++) {
        uint64_t dsto = __shfl(my_dst, r);
        uint64_t h0 = __shfl(my_h0, r);
        uint64_t h1 = __shfl(my_h1, r);
        uint32_t len = __shfl(my_len, r);
        uint32_t hdr = __shfl(my_hdr, r);
        uint32_t ph = __shfl(my_ph, r);
        uint8_t* w = out + dsto;
        if (lane < hdr)
          w[lane] = (lane < 8) ? (uint8_t)(h0 >> (8 * lane))
                               : (uint8_t)(h1 >> (8 * (lane - 8)));
        w += hdr;
        const uint8_t* ls = (const uint8_t*)stage[wv][r] + ph;
        for (uint32_t b = lane; b < len; b += WAVE) w[b] = ls[b];
      }
    } else {
      for (uint32_t r = 0; r < nvalid; r++) {
        uint64_t src = __shfl(my_src, r);
        uint64_t dsto = __shfl(my_dst, r);
        uint64_t h0 = __shfl(my_h0, r);
        uint64_t h1 = __shfl(my_h1, r);
        uint32_t len = __shfl(my_len, r);
        uint32_t hdr = __shfl(my_hdr, r);
        uint8_t* w = out + dsto;
        if (lane < hdr)
          w[lane] = (lane < 8) ? (uint8_t)(h0 >> (8 * lane))
                               : (uint8_t)(h1 >> (8 * (lane - 8)));
        w += hdr;
        const uint8_t* sp2 = (const uint8_t*)(uintptr_t)src;
        for (uint32_t b = lane; b < len; b += WAVE) w[b] = sp2[b];
      }
    }
  }
}

/* Staged-span emit: each WAVE owns 64 consecutive sorted records — within a
 * partition their output bytes are one contiguous span.  Every lane stages
 * its own record (header regs + funneled u32 window loads) into a wave-local
 * LDS image laid out so image words map to 4-aligned OUTPUT words, then the
 * wave stores the span as dense aligned u32s.  Wave-synchronous (no
 * barriers); LDS byte writes are byte-granular so record boundaries need no
 * read-modify-write.  Cross-partition / short / oversized waves fall back to
 * the per-record loop.  Fixes the two measured failure modes: scattered
 * partial-line stores (lane-per-record v1) and per-rotate register pressure
 * (word-funnel v2), while keeping 64 records of gather MLP in flight. */
#define SPAN_BYTES (64 * 144 + 16)
__global__ __launch_bounds__(BLOCK) void k_emit_span(
    const RecDesc* desc, const uint8_t* same, const uint64_t* scan,
    const uint32_t* parts, const uint64_t* seg_payload_start,
    const uint64_t* part_scan_base, uint8_t* out, uint32_t n) {
  __shared__ __attribute__((aligned(16))) uint8_t img8[WPB][SPAN_BYTES];
  uint32_t wave = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  uint32_t lane = threadIdx.x & (WAVE - 1);
  uint32_t wv = threadIdx.x / WAVE;
  uint32_t nwaves = (gridDim.x * blockDim.x) / WAVE;
  uint8_t* img = img8[wv];
  uint32_t* img32 = (uint32_t*)img;
  for (uint64_t base = (uint64_t)wave * WAVE; base < n;
       base += (uint64_t)nwaves * WAVE) {
    uint32_t i = (uint32_t)base + lane;
    uint64_t my_src = 0, my_dst = 0, my_h0 = 0, my_h1 = 0;
    uint32_t my_len = 0, my_hdr = 0, my_p = 0;
    if (i < n) {
      RecDesc v = desc[i];
      my_p = parts[i];
      my_dst = seg_payload_start[my_p] + (scan[i] - part_scan_base[my_p]);
      uint8_t prev_same = (i > 0 && parts[i] == parts[i - 1]) ? same[i - 1] : 0;
      uint8_t hdrbuf[16] = {0};
      uint32_t hdr = 0;
      if (same[i]) {
        if (!prev_same) hdrbuf[hdr++] = 0xFE;
        hdr += d_vint_write(hdrbuf + hdr, v.vlen);
        my_src = v.src + v.klen;
        my_len = v.vlen;
      } else {
        if (prev_same) hdrbuf[hdr++] = 0xFD;
        hdr += d_vint_write(hdrbuf + hdr, v.klen);
        hdr += d_vint_write(hdrbuf + hdr, v.vlen);
        my_src = v.src;
        my_len = v.klen + v.vlen;
      }
      my_hdr = hdr;
      for (int b = 0; b < 8; b++) my_h0 |= (uint64_t)hdrbuf[b] << (8 * b);
      for (int b = 8; b < 12; b++) my_h1 |= (uint64_t)hdrbuf[b] << (8 * (b - 8));
    }
    uint32_t nvalid = (n - base < WAVE) ? (uint32_t)(n - base) : WAVE;
    uint32_t p0 = __shfl(my_p, 0);
    uint64_t okb = __ballot(i >= n || my_p == p0);
    uint64_t span0 = __shfl(my_dst, 0);
    uint64_t span_end = __shfl(my_dst + my_hdr + my_len, WAVE - 1);
    uint32_t s = (uint32_t)(span0 & 3);
    uint64_t span_len = span_end - span0;
    if (nvalid == WAVE && okb == ~0ull && s + span_len <= SPAN_BYTES - 8) {
      uint32_t io = (uint32_t)(my_dst - span0) + s;
      uint32_t tot = my_hdr + my_len;
      const uint8_t* sp = (const uint8_t*)(uintptr_t)my_src;
      /* A: bytes up to the first aligned word boundary past the header */
      uint32_t B0 = (io + my_hdr + 3) & ~3u;
      if (B0 > io + tot) B0 = io + tot;
      for (uint32_t q = io; q < B0; q++) {
        uint32_t k = q - io;
        img[q] = (k < my_hdr) ? ((k < 8) ? (uint8_t)(my_h0 >> (8 * k))
                                         : (uint8_t)(my_h1 >> (8 * (k - 8))))
                              : sp[k - my_hdr];
      }
      /* B: full image words from funneled source window words, in batches of
         8 so the global loads issue independently (a serial prev/next funnel
         chain measured 3x slower: one vmcnt stall per word) */
      uint32_t B1 = (io + tot) & ~3u;
      if (B1 > B0) {
        uint32_t off0 = B0 - io - my_hdr;
        uint64_t S = my_src + off0;
        const uint32_t* sw = (const uint32_t*)(S & ~3ull);
        uint32_t sh = (uint32_t)(S & 3), nw = (B1 - B0) >> 2;
        uint32_t jw = B0 >> 2;
        uint32_t j = 0;
        while (j < nw) {
          uint32_t cnt = nw - j;
          if (cnt > 8) cnt = 8;
          uint32_t r[9];
          #pragma unroll
          for (uint32_t k = 0; k < 9; k++)
            if (k <= cnt) r[k] = sw[j + k];
          #pragma unroll
          for (uint32_t k = 0; k < 8; k++)
            if (k < cnt)
              img32[jw + j + k] =
                  sh ? ((r[k] >> (8 * sh)) | (r[k + 1] << (8 * (4 - sh)))) : r[k];
          j += cnt;
        }
      }
      /* C: tail bytes */
      for (uint32_t q = (B1 > B0 ? B1 : B0); q < io + tot; q++)
        img[q] = sp[q - io - my_hdr];
      /* wave store: head bytes, dense aligned words, tail bytes */
      uint32_t hb = (4 - s) & 3;
      if (hb > span_len) hb = (uint32_t)span_len;
      if (lane < hb) out[span0 + lane] = img[s + lane];
      uint32_t j0 = (s + hb) >> 2;
      uint32_t j1 = (uint32_t)((s + span_len) >> 2);
      for (uint32_t j = j0 + lane; j < j1; j += WAVE)
        *(uint32_t*)(out + span0 - s + 4ull * j) = img32[j];
      uint32_t tpos = 4u * j1;
      uint32_t tend = (uint32_t)(s + span_len);
      if (lane < tend - tpos) out[span0 - s + tpos + lane] = img[tpos + lane];
    } else {
      /* fallback: one record at a time, whole wave cooperates */
      for (uint32_t r = 0; r < nvalid; r++) {
        uint64_t src = __shfl(my_src, r);
        uint64_t dsto = __shfl(my_dst, r);
        uint64_t h0 = __shfl(my_h0, r);
        uint64_t h1 = __shfl(my_h1, r);
        uint32_t len = __shfl(my_len, r);
        uint32_t hdr = __shfl(my_hdr, r);
        uint8_t* w = out + dsto;
        if (lane < hdr)
          w[lane] = (lane < 8) ? (uint8_t)(h0 >> (8 * lane))
                               : (uint8_t)(h1 >> (8 * (lane - 8)));
        w += hdr;
        const uint8_t* sp2 = (const uint8_t*)(uintptr_t)src;
        for (uint32_t b = lane; b < len; b += WAVE) w[b] = sp2[b];
      }
    }
  }
}

/* Arithmetic uniform emit: single-segment uniform records with RLE off —
 * src/dst are pure arithmetic of (sidx, parts), the header word is a
 * constant, so the half-wave rotate needs only TWO shuffles per record and
 * the host skips the descriptor build and the size scan entirely
 * (~4.8 GB of traffic at C2). */
__global__ __launch_bounds__(BLOCK) void k_emit_uniform_arith(
    const uint8_t* data0, const uint32_t* sidx, const uint32_t* parts,
    const uint64_t* pstart /* [P+1] record ranges */,
    const uint64_t* seg_payload_start, uint8_t* out, uint32_t n,
    uint32_t rec_u, uint32_t hdr_len, uint64_t hdr_word) {
  constexpr int HB = 4;
  const uint32_t out_stride = hdr_len + rec_u;
  uint32_t wave = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  uint32_t lane = threadIdx.x & (WAVE - 1);
  uint32_t half = lane >> 5;
  uint32_t hl = lane & 31;
  uint32_t nwaves = (gridDim.x * blockDim.x) / WAVE;
  for (uint64_t base = (uint64_t)wave * WAVE; base < n;
       base += (uint64_t)nwaves * WAVE) {
    uint32_t i = (uint32_t)base + lane;
    uint64_t my_src = 0, my_dst = 0;
    if (i < n) {
      uint32_t p = parts[i];
      my_dst = seg_payload_start[p] + (i - pstart[p]) * out_stride;
      my_src = (uint64_t)(uintptr_t)(data0 + (uint64_t)sidx[i] * rec_u);
    }
    uint32_t nvalid = (n - base < WAVE) ? (uint32_t)(n - base) : WAVE;
    if (nvalid == WAVE && rec_u <= 32 * HB) {
      uint32_t r0 = half;
      uint64_t src0 = __shfl(my_src, r0), dst0 = __shfl(my_dst, r0);
      uint8_t b0[HB];
      {
        const uint8_t* sp = (const uint8_t*)(uintptr_t)src0;
#pragma unroll
        for (int k = 0; k < HB; k++)
          b0[k] = (hl + k * 32 < rec_u) ? sp[hl + k * 32] : 0;
      }
      for (uint32_t r = 0; r < 32; r++) {
        uint32_t rn = 2 * (r + 1) + half;
        uint64_t src1 = 0, dst1 = 0;
        uint8_t b1[HB] = {0};
        if (rn < WAVE) {
          src1 = __shfl(my_src, rn);
          dst1 = __shfl(my_dst, rn);
          const uint8_t* sp = (const uint8_t*)(uintptr_t)src1;
#pragma unroll
          for (int k = 0; k < HB; k++)
            b1[k] = (hl + k * 32 < rec_u) ? sp[hl + k * 32] : 0;
        }
        uint8_t* w = out + dst0;
        if (hl < hdr_len) w[hl] = (uint8_t)(hdr_word >> (8 * hl));
        w += hdr_len;
#pragma unroll
        for (int k = 0; k < HB; k++)
          if (hl + k * 32 < rec_u) w[hl + k * 32] = b0[k];
        src0 = src1; dst0 = dst1;
#pragma unroll
        for (int k = 0; k < HB; k++) b0[k] = b1[k];
      }
    } else {
      for (uint32_t r = 0; r < nvalid; r++) {
        uint64_t src = __shfl(my_src, r);
        uint64_t dsto = __shfl(my_dst, r);
        uint8_t* w = out + dsto;
        if (lane < hdr_len) w[lane] = (uint8_t)(hdr_word >> (8 * lane));
        w += hdr_len;
        const uint8_t* sp2 = (const uint8_t*)(uintptr_t)src;
        for (uint32_t b = lane; b < rec_u; b += WAVE) w[b] = sp2[b];
      }
    }
  }
}

/* Uniform-record emit fast path: when every record serializes to the same
 * length and RLE is off, the IFile body is a constant-stride stream
 * (hdr vints ‖ key ‖ val per record).  One LANE per record: the 88-byte
 * C2 record moves as eleven u64 loads and ~12 u64 stores (head/tail bytes
 * for output alignment) instead of the generic path's byte-granular
 * half-wave gather — ~20x fewer issue slots, dense aggregate writes. */
template <typename WordT, int MAXW>
__global__ __launch_bounds__(BLOCK) void k_emit_uniform(
    const RecDesc* desc, const uint64_t* scan, const uint32_t* parts,
    const uint64_t* seg_payload_start, const uint64_t* part_scan_base,
    uint8_t* out, uint32_t n, uint32_t hdrlen /* <= 8 */, uint64_t hdrword,
    uint32_t reclen /* = klen+vlen; % sizeof(WordT) == 0; <= MAXW words */) {
  constexpr uint32_t WB = (uint32_t)sizeof(WordT);
  const uint32_t total = hdrlen + reclen;
  const uint32_t nwords = reclen / WB;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x) {
    uint32_t p = parts[i];
    uint64_t dst = seg_payload_start[p] + (scan[i] - part_scan_base[p]);
    const WordT* srcw = (const WordT*)(uintptr_t)desc[i].src;
    WordT s[MAXW + 1];
    #pragma unroll
    for (uint32_t k = 0; k < MAXW + 1; k++) s[k] = (k < nwords) ? srcw[k] : 0;
    /* q0: first logical offset >= hdrlen that is WB-aligned in OUTPUT */
    uint32_t a = (uint32_t)(dst % WB);
    uint32_t q0 = hdrlen + ((WB - ((a + hdrlen) % WB)) % WB);
    for (uint32_t q = 0; q < q0 && q < total; q++) {
      uint8_t b = (q < hdrlen) ? (uint8_t)(hdrword >> (8 * q))
                               : (uint8_t)(s[0] >> (8 * (q - hdrlen)));
      out[dst + q] = b;
    }
    /* aligned word body: word k covers logical [q0 + WB*k, +WB) */
    uint32_t c = (q0 - hdrlen) % WB;
    uint32_t kmax = (total >= q0 + WB) ? (total - q0) / WB : 0;
    WordT* ww = (WordT*)(out + dst + q0);
    if (c == 0) {
      for (uint32_t k = 0; k < kmax; k++) ww[k] = s[k];
    } else {
      for (uint32_t k = 0; k < kmax; k++)
        ww[k] = (WordT)((s[k] >> (8 * c)) | (s[k + 1] << (8 * (WB - c))));
    }
    for (uint32_t q = q0 + WB * kmax; q < total; q++) {
      uint32_t o = q - hdrlen;
      out[dst + q] = (uint8_t)(s[o / WB] >> (8 * (o % WB)));
    }
  }
}

/* permuted-columnar materialization (exchange wire — DESIGN.md §4):
 * records gathered into sorted order as (data, off, klen) so partition
 * ranges are contiguous per destination rank. */
__global__ void k_sorted_reclens(RecTable rt, const uint32_t* sidx, uint64_t* lens,
                                 uint32_t n) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x) {
    RecView v = rt_view(rt, sidx[i]);
    lens[i] = (uint64_t)v.klen + v.vlen;
  }
}
__global__ void k_permute_records(RecTable rt, const uint32_t* sidx,
                                  const uint64_t* out_off, uint8_t* out_data,
                                  uint32_t* out_klen, uint32_t n) {
  uint32_t wave = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  uint32_t lane = threadIdx.x & (WAVE - 1);
  uint32_t nwaves = (gridDim.x * blockDim.x) / WAVE;
  for (uint32_t i = wave; i < n; i += nwaves) {
    RecView v = rt_view(rt, sidx[i]);
    uint8_t* w = out_data + out_off[i];
    uint32_t len = v.klen + v.vlen;  /* key ‖ val contiguous */
    for (uint32_t b = lane; b < len; b += WAVE) w[b] = v.key[b];
    if (lane == 0) out_klen[i] = v.klen;
  }
}

/* ---- CRC over emitted segments ----
 * chunk kernel: thread computes CRC32 of one 256-byte chunk of one partition's
 * checksummed range (payload .. EOF); combine kernel: one wave per partition
 * reduces its chunk list sequentially with the 256-byte shift matrix. */
#define CRC_CHUNK 256
__constant__ uint32_t c_crc_table[256];
__constant__ uint32_t c_crc_table8[8][256]; /* slice-by-8: T0 = standard */
__constant__ uint32_t c_crc_mats[CRC_MATS][32];
/* byte-sliced SHIFT operators (multiply by x^(8*64) / x^(8*256)):
 * shiftN(crc) = T[0][crc&FF] ^ T[1][crc>>8 &FF] ^ T[2][..] ^ T[3][crc>>24]
 * — 4 table lookups instead of a 32-step serial GF(2) matrix apply. */
__constant__ uint32_t c_crc_t64[4][256];
__constant__ uint32_t c_crc_t256[4][256];

__device__ __forceinline__ uint32_t d_crc_shift(uint32_t crc, uint64_t nbytes) {
  for (int k = 0; nbytes; k++, nbytes >>= 1)
    if (nbytes & 1) {
      uint32_t s = 0, v = crc;
      for (int i = 0; v; i++, v >>= 1)
        if (v & 1) s ^= c_crc_mats[k][i];
      crc = s;
    }
  return crc;
}

/* Super-chunk staged CRC v3: one block stages 128 chunks (32 KB) of a
 * partition's checksummed range into LDS via coalesced aligned-u32 loads
 * with a funnel shift for misaligned range starts, then each thread CRCs its
 * 256-byte chunk from LDS with slice-by-4 tables (4 lookups per word, 1/4 the
 * dependent-chain length of bytewise).  Rows padded one word so lane t's
 * reads land on distinct banks. */
#define CRC_SC_CHUNKS 64
#define CRC_SC_BYTES (CRC_SC_CHUNKS * CRC_CHUNK) /* 16 KiB per super-chunk */
/* Register-direct chunk CRC: 4 threads per 256B chunk, one 64B quarter each.
 * Each lane loads its quarter straight from the stream into 17 registers
 * (4-aligned u32 + funnel for the range's byte phase; the word after a range
 * always lands inside that range's 4B CRC trailer, so the +1 read is in
 * bounds), runs an 8-step slice-by-8 chain from registers, and the three
 * shuffle-gathered quarter CRCs merge with the byte-sliced shift-by-64
 * operator.  No data staging, no per-iteration barriers — the old LDS-staged
 * version was 84% wave-parked (PMC, profiles/) on the load->sync->compute
 * serialization. */
__global__ __launch_bounds__(BLOCK) void k_crc_chunks(
    const uint8_t* stream, const uint64_t* range_start, const uint64_t* range_len,
    const uint64_t* chunk_base, const uint64_t* sc_base /* [P+1] */,
    uint32_t nparts, uint32_t total_sc, uint32_t* chunk_crc) {
  __shared__ uint32_t tab8[8][256];
  __shared__ uint32_t t64[4][256];
  for (int i = threadIdx.x; i < 2048; i += blockDim.x)
    ((uint32_t*)tab8)[i] = ((const uint32_t*)c_crc_table8)[i];
  for (int i = threadIdx.x; i < 1024; i += blockDim.x)
    ((uint32_t*)t64)[i] = ((const uint32_t*)c_crc_t64)[i];
  __syncthreads();
  const uint32_t lane = threadIdx.x & (WAVE - 1);
  for (uint32_t sc = blockIdx.x; sc < total_sc; sc += gridDim.x) {
    uint32_t lo = 0, hi = nparts;
    while (lo + 1 < hi) {
      uint32_t mid = (lo + hi) / 2;
      if (sc_base[mid] <= sc) lo = mid; else hi = mid;
    }
    uint32_t p = lo;
    uint64_t sc_local = sc - sc_base[p];
    uint64_t byte0 = sc_local * CRC_SC_BYTES;
    uint64_t avail = range_len[p] - byte0;
    if (avail > CRC_SC_BYTES) avail = CRC_SC_BYTES;
    uint64_t gbase = range_start[p] + byte0;
    uint32_t a = (uint32_t)(gbase & 3);
    const uint32_t* wsrc = (const uint32_t*)(stream + gbase - a);
    uint32_t chunk = threadIdx.x >> 2, quarter = threadIdx.x & 3;
    uint64_t c0 = (uint64_t)chunk * CRC_CHUNK;
    bool live = c0 < avail;
    uint64_t len = live ? min(avail - c0, (uint64_t)CRC_CHUNK) : 0;
    uint32_t crc = 0;
    if (live && len == CRC_CHUNK) {
      uint32_t j0 = ((uint32_t)c0 >> 2) + quarter * 16;
      uint32_t r[17];
      #pragma unroll
      for (int k = 0; k < 17; k++) r[k] = wsrc[j0 + k];
      crc = (quarter == 0) ? 0xFFFFFFFFu : 0u;
      if (a == 0) {
        #pragma unroll
        for (uint32_t w = 0; w < 8; w++) {
          uint32_t lo32 = crc ^ r[2 * w];
          uint32_t hi32 = r[2 * w + 1];
          crc = tab8[7][lo32 & 0xFF] ^ tab8[6][(lo32 >> 8) & 0xFF]
              ^ tab8[5][(lo32 >> 16) & 0xFF] ^ tab8[4][lo32 >> 24]
              ^ tab8[3][hi32 & 0xFF] ^ tab8[2][(hi32 >> 8) & 0xFF]
              ^ tab8[1][(hi32 >> 16) & 0xFF] ^ tab8[0][hi32 >> 24];
        }
      } else {
        uint32_t sh = 8 * a, ish = 32 - sh;
        #pragma unroll
        for (uint32_t w = 0; w < 8; w++) {
          uint32_t lo32 = crc ^ ((r[2 * w] >> sh) | (r[2 * w + 1] << ish));
          uint32_t hi32 = (r[2 * w + 1] >> sh) | (r[2 * w + 2] << ish);
          crc = tab8[7][lo32 & 0xFF] ^ tab8[6][(lo32 >> 8) & 0xFF]
              ^ tab8[5][(lo32 >> 16) & 0xFF] ^ tab8[4][lo32 >> 24]
              ^ tab8[3][hi32 & 0xFF] ^ tab8[2][(hi32 >> 8) & 0xFF]
              ^ tab8[1][(hi32 >> 16) & 0xFF] ^ tab8[0][hi32 >> 24];
        }
      }
    } else if (live && quarter == 0) {
      /* ragged tail chunk (only the last chunk of a partition): bytewise */
      const uint8_t* sp = stream + gbase + c0;
      crc = 0xFFFFFFFFu;
      for (uint64_t b = 0; b < len; b++)
        crc = (crc >> 8) ^ tab8[0][(crc ^ sp[b]) & 0xFF];
    }
    /* shuffle-combine the four quarter CRCs (quarters are adjacent lanes) */
    uint32_t qbase = lane & ~3u;
    uint32_t v1 = __shfl(crc, qbase + 1);
    uint32_t v2 = __shfl(crc, qbase + 2);
    uint32_t v3 = __shfl(crc, qbase + 3);
    if (live && quarter == 0) {
      if (len == CRC_CHUNK) {
        uint32_t vs[3] = {v1, v2, v3};
        #pragma unroll
        for (int q = 0; q < 3; q++) {
          crc = t64[0][crc & 0xFF] ^ t64[1][(crc >> 8) & 0xFF]
              ^ t64[2][(crc >> 16) & 0xFF] ^ t64[3][crc >> 24];
          crc ^= vs[q];
        }
      }
      chunk_crc[chunk_base[p] + sc_local * CRC_SC_CHUNKS + chunk] = crc ^ 0xFFFFFFFFu;
    }
  }
}

/* Tree combine, two levels: thread-serial fold of 8 chunks, then an LDS tree
 * over ADJACENT spans; a final kernel folds each partition's groups.  Every
 * fold is combine(L,R) = shift(crc_L, len_R) ^ crc_R with the generic
 * bit-decomposed shift so ragged tails are exact. */
#define CRC_GROUP_CHUNKS 2048
__global__ void k_crc_combine_groups(const uint64_t* range_len, const uint64_t* chunk_base,
                                     const uint64_t* group_base /* [P+1] */,
                                     const uint32_t* chunk_crc, uint32_t nparts,
                                     uint32_t total_groups,
                                     uint32_t* group_crc, uint64_t* group_len) {
  __shared__ uint32_t s_crc[BLOCK];
  __shared__ uint64_t s_len[BLOCK];
  __shared__ uint32_t t256[4][256];
  for (int i = threadIdx.x; i < 1024; i += blockDim.x)
    ((uint32_t*)t256)[i] = ((const uint32_t*)c_crc_t256)[i];
  __syncthreads();
  for (uint32_t g = blockIdx.x; g < total_groups; g += gridDim.x) {
    /* find partition via binary search over group_base */
    uint32_t lo = 0, hi = nparts;
    while (lo + 1 < hi) {
      uint32_t mid = (lo + hi) / 2;
      if (group_base[mid] <= g) lo = mid; else hi = mid;
    }
    uint32_t p = lo;
    uint64_t glocal = g - group_base[p];
    uint64_t len = range_len[p];
    uint64_t nchunks = (len + CRC_CHUNK - 1) / CRC_CHUNK;
    uint64_t c0 = glocal * CRC_GROUP_CHUNKS;
    uint64_t cend = min(c0 + (uint64_t)CRC_GROUP_CHUNKS, nchunks);
    /* thread-serial fold of 8 consecutive chunks */
    uint64_t t0 = c0 + (uint64_t)threadIdx.x * 8;
    uint32_t crc = 0;
    uint64_t mylen = 0;
    for (uint64_t c = t0; c < min(t0 + 8, cend); c++) {
      uint64_t clen = len - c * CRC_CHUNK;
      if (clen > CRC_CHUNK) clen = CRC_CHUNK;
      if (clen == CRC_CHUNK)
        crc = t256[0][crc & 0xFF] ^ t256[1][(crc >> 8) & 0xFF]
            ^ t256[2][(crc >> 16) & 0xFF] ^ t256[3][crc >> 24]
            ^ chunk_crc[chunk_base[p] + c];
      else
        crc = d_crc_shift(crc, clen) ^ chunk_crc[chunk_base[p] + c];
      mylen += clen;
    }
    s_crc[threadIdx.x] = crc;
    s_len[threadIdx.x] = mylen;
    __syncthreads();
    /* LDS tree over ADJACENT spans: CRC combine is associative but NOT
     * commutative, so thread t merges [t, t+stride) with [t+stride, t+2s). */
    for (int stride = 1; stride < BLOCK; stride <<= 1) {
      if ((threadIdx.x & (2 * stride - 1)) == 0 && threadIdx.x + stride < BLOCK) {
        uint64_t rl = s_len[threadIdx.x + stride];
        if (rl) {
          s_crc[threadIdx.x] = d_crc_shift(s_crc[threadIdx.x], rl)
                               ^ s_crc[threadIdx.x + stride];
          s_len[threadIdx.x] += rl;
        }
      }
      __syncthreads();
    }
    if (threadIdx.x == 0) { group_crc[g] = s_crc[0]; group_len[g] = s_len[0]; }
    __syncthreads();
  }
}
__global__ void k_crc_combine_final(const uint64_t* group_base, const uint32_t* group_crc,
                                    const uint64_t* group_len, uint32_t nparts,
                                    uint32_t* part_crc) {
  /* one WAVE per partition; each GF(2) matrix application is lane-parallel:
   * lane i contributes mats[k][i] when bit i of the crc is set, then a
   * butterfly XOR-reduces across the wave (6 steps instead of a 32-step
   * serial fold per shift). */
  uint32_t wave = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  uint32_t lane = threadIdx.x & (WAVE - 1);
  if (wave >= nparts) return;
  uint32_t p = wave;
  uint32_t crc = 0;
  for (uint64_t g = group_base[p]; g < group_base[p + 1]; g++) {
    uint64_t nb = group_len[g];
    if (!nb) continue;
    for (int k = 0; nb; k++, nb >>= 1) {
      if (nb & 1) {
        uint32_t contrib = (lane < 32 && ((crc >> lane) & 1)) ? c_crc_mats[k][lane] : 0;
        #pragma unroll
        for (int s = 32; s >= 1; s >>= 1) contrib ^= __shfl_xor(contrib, s, WAVE);
        crc = contrib;
      }
    }
    crc ^= group_crc[g];
  }
  if (lane == 0) part_crc[p] = crc;
}

/* write headers, EOF trailers and CRC for each partition segment */
__global__ void k_patch_segments(uint8_t* out, const uint64_t* seg_start,
                                 const uint64_t* body_len, const uint8_t* last_same,
                                 const uint32_t* part_crc, const uint8_t* seg_present,
                                 uint32_t nparts) {
  uint32_t p = blockIdx.x * blockDim.x + threadIdx.x;
  if (p >= nparts || !seg_present[p]) return;
  uint8_t* s = out + seg_start[p];
  s[0] = 'T'; s[1] = 'I'; s[2] = 'F'; s[3] = 0;
  uint8_t* e = s + 4 + body_len[p];
  if (last_same[p]) *e++ = 0xFD; /* V_END before EOF */
  *e++ = 0xFF; *e++ = 0xFF;      /* EOF -1 -1 */
  uint32_t crc = part_crc[p];
  e[0] = (uint8_t)(crc >> 24); e[1] = (uint8_t)(crc >> 16);
  e[2] = (uint8_t)(crc >> 8); e[3] = (uint8_t)crc;
}

/* ---- device deflate (TIF\\1 compressed IFile segments) ----------------
 * SURVEY 8f row 2 / VERDICT r1 #9: the reference compresses each IFile
 * segment with a Hadoop codec (DefaultCodec = one zlib stream per segment,
 * CRC32 over the COMPRESSED payload - IFile.java:352-368, pinned by the
 * golden fixture).  MI355X-native encoder: the payload splits into 32 KB
 * chunks; each chunk is deflated by one wave (greedy LZ77 over an LDS hash
 * table + fixed-Huffman emit in lane 0, lane-parallel stored-block copy
 * when incompressible) into its own slot, each chunk ending byte-aligned
 * via an empty stored (sync-flush) block so the slots concatenate into ONE
 * valid zlib stream; per-chunk adler32 halves combine on the host.
 * Compressed BYTES are encoder-specific (any inflater accepts them); the
 * reference contract is the framing + CRC + decompressed payload. */
#define DEF_CHUNK 32768
#define DEF_SLOT  (DEF_CHUNK + DEF_CHUNK / 8 + 64) /* fixed-huffman worst < 9/8 */
#define DEF_HASH_BITS 12
#define DEF_HASH (1 << DEF_HASH_BITS)

struct DefChunk {
  uint64_t in_off;   /* absolute offset in the input stream */
  uint32_t in_len;   /* <= DEF_CHUNK */
  uint32_t last;     /* 1 = final chunk of its segment (BFINAL) */
};

__device__ __forceinline__ uint32_t d_bitrev(uint32_t v, int n) {
  uint32_t r = 0;
  for (int i = 0; i < n; i++) { r = (r << 1) | (v & 1); v >>= 1; }
  return r;
}

/* length code: base lengths + extra bits (RFC1951 3.2.5) */
__constant__ uint16_t c_len_base[29] = {3,4,5,6,7,8,9,10,11,13,15,17,19,23,27,31,
                                        35,43,51,59,67,83,99,115,131,163,195,227,258};
__constant__ uint8_t c_len_extra[29] = {0,0,0,0,0,0,0,0,1,1,1,1,2,2,2,2,
                                        3,3,3,3,4,4,4,4,5,5,5,5,0};
__constant__ uint16_t c_dist_base[30] = {1,2,3,4,5,7,9,13,17,25,33,49,65,97,129,193,
                                         257,385,513,769,1025,1537,2049,3073,4097,6145,
                                         8193,12289,16385,24577};
__constant__ uint8_t c_dist_extra[30] = {0,0,0,0,1,1,2,2,3,3,4,4,5,5,6,6,
                                         7,7,8,8,9,9,10,10,11,11,12,12,13,13};

struct DefBitWriter {
  uint8_t* out;
  uint64_t acc;
  uint32_t nbits;
  uint32_t pos;
  __device__ void put(uint32_t bits, uint32_t n) {
    acc |= (uint64_t)bits << nbits;
    nbits += n;
    while (nbits >= 8) {
      out[pos++] = (uint8_t)acc;
      acc >>= 8;
      nbits -= 8;
    }
  }
  __device__ void align() {
    if (nbits) { out[pos++] = (uint8_t)acc; acc = 0; nbits = 0; }
  }
};

/* fixed-huffman literal/length code (already bit-reversed for LSB packing) */
__device__ __forceinline__ void d_fh_lit(DefBitWriter& bw, uint32_t v) {
  if (v < 144) bw.put(d_bitrev(0x30 + v, 8), 8);
  else bw.put(d_bitrev(0x190 + (v - 144), 9), 9);
}
__device__ __forceinline__ void d_fh_len(DefBitWriter& bw, uint32_t len) {
  int c = 28;
  while (c > 0 && c_len_base[c] > len) c--;
  /* code 257+c: 257..279 -> 7 bits (code-256), 280..285 -> 8 bits */
  uint32_t sym = 257 + c;
  if (sym < 280) bw.put(d_bitrev(sym - 256, 7), 7);
  else bw.put(d_bitrev(0xC0 + (sym - 280), 8), 8);
  if (c_len_extra[c]) bw.put(len - c_len_base[c], c_len_extra[c]);
}
__device__ __forceinline__ void d_fh_dist(DefBitWriter& bw, uint32_t dist) {
  int c = 29;
  while (c > 0 && c_dist_base[c] > dist) c--;
  bw.put(d_bitrev((uint32_t)c, 5), 5);
  if (c_dist_extra[c]) bw.put(dist - c_dist_base[c], c_dist_extra[c]);
}

/* one wave per chunk; lane 0 runs the serial parse/emit, all lanes build
 * the hash table and handle the stored-block fallback copy */
__global__ __launch_bounds__(WAVE) void k_deflate_chunks(
    const uint8_t* in, const DefChunk* chunks, uint32_t nchunks,
    uint8_t* slots /* [nchunks][DEF_SLOT] */, uint32_t* out_len,
    uint32_t* out_adler /* [nchunks][2]: a, b */) {
  __shared__ uint16_t htab[DEF_HASH];
  const uint32_t lane = threadIdx.x;
  for (uint32_t c = blockIdx.x; c < nchunks; c += gridDim.x) {
    DefChunk ck = chunks[c];
    const uint8_t* src = in + ck.in_off;
    uint8_t* dst = slots + (uint64_t)c * DEF_SLOT;
    for (uint32_t i = lane; i < DEF_HASH; i += WAVE) htab[i] = 0xFFFF;
    __syncthreads();  /* one wave per block: cheap */
    /* adler32 halves: lane-parallel partial sums then lane-0 fold.
       a = 1 + S bytes; b = len + S (len-i)*byte_i  (mod 65521) */
    uint64_t sa = 0, sb = 0;
    for (uint32_t i = lane; i < ck.in_len; i += WAVE) {
      uint32_t v = src[i];
      sa += v;
      sb += (uint64_t)v * (ck.in_len - i);
    }
    for (int sh = 32; sh >= 1; sh >>= 1) {
      sa += __shfl_xor(sa, sh);
      sb += __shfl_xor(sb, sh);
    }
    if (lane == 0) {
      out_adler[2 * c] = (uint32_t)((1 + sa) % 65521u);
      out_adler[2 * c + 1] = (uint32_t)((ck.in_len + sb) % 65521u);
      /* serial greedy LZ77 + fixed-huffman */
      DefBitWriter bw;
      bw.out = dst;
      bw.acc = 0; bw.nbits = 0; bw.pos = 0;
      bw.put(0, 1);      /* BFINAL=0 (the closing stored block carries it) */
      bw.put(1, 2);      /* BTYPE=01 fixed huffman */
      uint32_t n2 = ck.in_len;
      uint32_t i = 0;
      uint32_t budget = DEF_CHUNK + DEF_CHUNK / 16; /* abort to stored */
      while (i < n2) {
        if (bw.pos >= budget) break;
        uint32_t mlen = 0, mdist = 0;
        if (i + 3 <= n2) {
          uint32_t w3 = (uint32_t)src[i] | ((uint32_t)src[i + 1] << 8) |
                        ((uint32_t)src[i + 2] << 16);
          uint32_t h = (w3 * 2654435761u) >> (32 - DEF_HASH_BITS);
          uint32_t cand = htab[h];
          htab[h] = (uint16_t)i;
          if (cand != 0xFFFF && cand < i) {
            const uint8_t* p = src + cand;
            const uint8_t* q = src + i;
            uint32_t maxm = n2 - i;
            if (maxm > 258) maxm = 258;
            uint32_t l = 0;
            while (l < maxm && p[l] == q[l]) l++;
            if (l >= 3) { mlen = l; mdist = i - cand; }
          }
        }
        if (mlen) {
          d_fh_len(bw, mlen);
          d_fh_dist(bw, mdist);
          /* seed the table inside the match (sparsely: every 2nd byte) */
          uint32_t e = i + mlen;
          for (uint32_t j = i + 1; j + 3 <= n2 && j < e; j += 2) {
            uint32_t w3 = (uint32_t)src[j] | ((uint32_t)src[j + 1] << 8) |
                          ((uint32_t)src[j + 2] << 16);
            htab[(w3 * 2654435761u) >> (32 - DEF_HASH_BITS)] = (uint16_t)j;
          }
          i = e;
        } else {
          d_fh_lit(bw, src[i]);
          i++;
        }
      }
      if (i >= n2) {
        bw.put(0, 7);  /* end-of-block (code 256, 7 zero bits) */
        /* closing empty stored block (sync flush) carries this chunk's
           BFINAL; its 3 header bits continue the BIT stream, THEN the
           stream pads to a byte boundary, then LEN/NLEN — every chunk
           ends byte-aligned so the slots concatenate */
        bw.put(ck.last ? 1u : 0u, 1);
        bw.put(0, 2); /* BTYPE=00 */
        bw.align();
        bw.out[bw.pos++] = 0x00;
        bw.out[bw.pos++] = 0x00;
        bw.out[bw.pos++] = 0xFF;
        bw.out[bw.pos++] = 0xFF;
        out_len[c] = bw.pos;
      } else {
        out_len[c] = 0xFFFFFFFFu; /* sentinel: use stored */
      }
    }
    __syncthreads();
    /* broadcast lane 0's decision; stored fallback copies with all lanes */
    uint32_t marker = (uint32_t)__shfl(lane == 0 ? (int)out_len[c] : 0, 0);
    if (marker == 0xFFFFFFFFu) {
      /* stored block: BFINAL=0,BTYPE=00 (1 byte 0x00), LEN, NLEN, raw;
         data ends byte-aligned, so the closing block header starts a new
         byte here */
      if (lane == 0) {
        dst[0] = 0x00;
        dst[1] = (uint8_t)ck.in_len;
        dst[2] = (uint8_t)(ck.in_len >> 8);
        dst[3] = (uint8_t)~dst[1];
        dst[4] = (uint8_t)~dst[2];
        uint32_t clen = 5 + ck.in_len;
        dst[clen] = ck.last ? 0x01 : 0x00;
        dst[clen + 1] = 0x00;
        dst[clen + 2] = 0x00;
        dst[clen + 3] = 0xFF;
        dst[clen + 4] = 0xFF;
        out_len[c] = clen + 5;
      }
      for (uint32_t i = lane; i < ck.in_len; i += WAVE) dst[5 + i] = src[i];
    }
  }
}

/* compact chunk slots into the final stream at host-computed offsets */
__global__ void k_deflate_gather(const uint8_t* slots, const uint32_t* lens,
                                 const uint64_t* dst_off, uint32_t nchunks,
                                 uint8_t* out) {
  uint32_t wave = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  uint32_t lane = threadIdx.x & (WAVE - 1);
  uint32_t nwaves = (gridDim.x * blockDim.x) / WAVE;
  for (uint32_t c = wave; c < nchunks; c += nwaves) {
    const uint8_t* s = slots + (uint64_t)c * DEF_SLOT;
    uint8_t* d = out + dst_off[c];
    uint32_t n = lens[c];
    for (uint32_t i = lane; i < n; i += WAVE) d[i] = s[i];
  }
}

/* ---- synthetic generation (bench/tests) ---- */
__device__ __forceinline__ uint64_t d_splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}
/* kind 0 (C2/C5 shape): BytesWritable keys of klen random bytes (uniqueness by
 * mixing record id), values vlen bytes. record = [4B BE klen][key][4B BE vlen][val] */
__global__ void k_generate_fixed(uint64_t seed, int64_t n, int32_t klen, int32_t vlen,
                                 uint8_t* data, uint64_t rec_bytes) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint8_t* r = data + (uint64_t)i * rec_bytes;
    r[0] = (uint8_t)(klen >> 24); r[1] = (uint8_t)(klen >> 16);
    r[2] = (uint8_t)(klen >> 8); r[3] = (uint8_t)klen;
    uint64_t s = seed ^ (uint64_t)i * 0x9E3779B97F4A7C15ull;
    int32_t b = 0;
    uint64_t w = 0;
    for (; b < klen; b++) {
      if ((b & 7) == 0) w = d_splitmix64(s + 1 + (b >> 3));
      uint8_t byte = (uint8_t)(w >> (8 * (b & 7)));
      /* mix the record id into bytes 4..8 to force uniqueness */
      if (b >= 4 && b < 8) byte ^= (uint8_t)((uint64_t)i >> (8 * (b - 4)));
      r[4 + b] = byte;
    }
    uint8_t* v = r + 4 + klen;
    v[0] = (uint8_t)(vlen >> 24); v[1] = (uint8_t)(vlen >> 16);
    v[2] = (uint8_t)(vlen >> 8); v[3] = (uint8_t)vlen;
    for (int32_t c = 0; c < vlen; c += 8) {
      uint64_t wv = d_splitmix64(s ^ 0x5bf03635ull ^ (uint64_t)(c >> 3));
      for (int32_t k2 = 0; k2 < 8 && c + k2 < vlen; k2++)
        v[4 + c + k2] = (uint8_t)(wv >> (8 * k2));
    }
  }
}
/* kind 2 (C5, TeraSort-shaped): 10B key + 90B value, range partitions
 * (TotalOrderPartitioner-style splits on the leading 2 key bytes). */
__global__ void k_range_partition(const uint8_t* data, uint64_t rec_bytes, int32_t P,
                                  int32_t* d_part, int64_t n) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const uint8_t* k = data + (uint64_t)i * rec_bytes + 4;
    uint32_t v = ((uint32_t)k[0] << 8) | k[1];
    d_part[i] = (int32_t)((uint64_t)v * (uint32_t)P >> 16);
  }
}
/* kind 3 (C4): uniform keys mapped through a host-built inverse-CDF LUT so
 * PARTITION SIZES follow Zipf(1.0) — BASELINE configs[3]'s skewed
 * 199-partition all-to-all-v shape. */
__global__ void k_lut_partition(const uint8_t* data, uint64_t rec_bytes,
                                const int32_t* lut /* [65536] */,
                                int32_t* d_part, int64_t n) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const uint8_t* k = data + (uint64_t)i * rec_bytes + 4;
    uint32_t v = ((uint32_t)k[0] << 8) | k[1];
    d_part[i] = lut[v];
  }
}

/* kind 1 (C3): Text keys, content = zipf(1.1)-drawn dict word + unique
 * base-36 record-id suffix, total length in [4,32]; 64B BytesWritable
 * values.  Lengths are computed twice (len pass + fill pass) from the same
 * deterministic draw. */
__device__ __forceinline__ void d_c3_key(uint64_t seed, int64_t i,
                                         uint8_t* out /* <=32 */, int* out_len) {
  uint64_t r = d_splitmix64(seed ^ (uint64_t)i * 0x9E3779B97F4A7C15ull);
  /* zipf-ish rank via inverse tail cdf u^(-1/(s-1)), s=1.1, clamped to 1e6 */
  double u = ((double)(r >> 11) + 1.0) / 9007199254740992.0;
  double z = exp(-10.0 * log(u));
  uint32_t word = (z >= 1e6) ? 999999u : (uint32_t)z;
  uint64_t wh = d_splitmix64(0xC3C3ull ^ word);
  int wl = 3 + (int)(wh % 8); /* 3..10 word chars */
  int n = 0;
  for (; n < wl; n++) out[n] = 'a' + (uint8_t)((wh >> (5 * (n % 12))) % 26);
  /* unique suffix: '#' + base36(i) */
  out[n++] = '#';
  uint64_t v = (uint64_t)i;
  uint8_t tmp[14];
  int t = 0;
  do { uint64_t d = v % 36; tmp[t++] = (uint8_t)(d < 10 ? '0' + d : 'a' + d - 10); v /= 36; } while (v);
  while (t && n < 32) out[n++] = tmp[--t];
  if (n < 4) { while (n < 4) out[n++] = '_'; }
  *out_len = n;
}
__global__ void k_gen_text_lens(uint64_t seed, int64_t n, int32_t vlen, uint64_t* lens) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint8_t kb[32];
    int kl;
    d_c3_key(seed, i, kb, &kl);
    /* Text key: 1-byte vint (len<=32) + content; value: 4B BE len + vlen */
    lens[i] = (uint64_t)(1 + kl) + 4 + (uint64_t)vlen;
  }
}
__global__ void k_gen_text_fill(uint64_t seed, int64_t n, int32_t vlen,
                                const uint64_t* off, uint8_t* data, uint32_t* klen_arr) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint8_t kb[32];
    int kl;
    d_c3_key(seed, i, kb, &kl);
    uint8_t* r = data + off[i];
    r[0] = (uint8_t)kl; /* vint: 0<len<=32 is a single byte */
    for (int b = 0; b < kl; b++) r[1 + b] = kb[b];
    uint8_t* v = r + 1 + kl;
    v[0] = (uint8_t)(vlen >> 24); v[1] = (uint8_t)(vlen >> 16);
    v[2] = (uint8_t)(vlen >> 8); v[3] = (uint8_t)vlen;
    uint64_t sv = d_splitmix64(seed ^ 0x77ull ^ (uint64_t)i);
    for (int32_t c = 0; c < vlen; c += 8) {
      uint64_t w = d_splitmix64(sv + (uint64_t)(c >> 3));
      for (int32_t k2 = 0; k2 < 8 && c + k2 < vlen; k2++)
        v[4 + c + k2] = (uint8_t)(w >> (8 * k2));
    }
    klen_arr[i] = (uint32_t)(1 + kl);
  }
}

__global__ void k_fill_fixed_offsets(uint64_t* off, uint32_t* klen_arr, int64_t n,
                                     uint64_t rec_bytes, uint32_t klen_ser) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i <= n;
       i += (int64_t)gridDim.x * blockDim.x) {
    off[i] = (uint64_t)i * rec_bytes;
    if (i < n) klen_arr[i] = klen_ser;
  }
}

/* ================================================================== */
/* host-side engine                                                    */
/* ================================================================== */
namespace {

struct DeviceInit {
  int inited = 0;
};

static int ensure_device_constants() {
  static bool done = false;
  if (done) return 0;
  h_crc_init();
  h_build_crc_mats();
  HIP_CHECK(hipMemcpyToSymbol(HIP_SYMBOL(c_crc_table), h_crc_table, sizeof(h_crc_table)));
  {
    static uint32_t t8[8][256];
    for (int i = 0; i < 256; i++) t8[0][i] = h_crc_table[i];
    for (int k = 1; k < 8; k++)
      for (int i = 0; i < 256; i++)
        t8[k][i] = (t8[k - 1][i] >> 8) ^ h_crc_table[t8[k - 1][i] & 0xFF];
    HIP_CHECK(hipMemcpyToSymbol(HIP_SYMBOL(c_crc_table8), t8, sizeof(t8)));
  }
  HIP_CHECK(hipMemcpyToSymbol(HIP_SYMBOL(c_crc_mats), h_crc_shift_mat, sizeof(h_crc_shift_mat)));
  {
    /* byte-sliced shift-by-64 / shift-by-256 operators:
     * T[j][b] = (x^(8*N)) * (b << 8j) mod P — so shiftN(crc) is 4 lookups */
    static uint32_t t64[4][256], t256[4][256];
    for (int j = 0; j < 4; j++)
      for (int b = 0; b < 256; b++) {
        t64[j][b] = gf2_times(h_crc_shift_mat[6], (uint32_t)b << (8 * j));
        t256[j][b] = gf2_times(h_crc_shift_mat[8], (uint32_t)b << (8 * j));
      }
    HIP_CHECK(hipMemcpyToSymbol(HIP_SYMBOL(c_crc_t64), t64, sizeof(t64)));
    HIP_CHECK(hipMemcpyToSymbol(HIP_SYMBOL(c_crc_t256), t256, sizeof(t256)));
  }
  done = true;
  return 0;
}

static inline uint32_t nblocks_for(uint64_t n, uint32_t per) {
  uint64_t b = (n + per - 1) / per;
  if (b == 0) b = 1;
  return (uint32_t)b;
}
static inline uint32_t grid1d(uint64_t n) {
  uint64_t b = (n + BLOCK - 1) / BLOCK;
  if (b > 2048) b = 2048;
  if (b == 0) b = 1;
  return (uint32_t)b;
}
/* wave-per-record kernels are latency-bound on per-record setup loads:
 * give them more waves */
static inline uint32_t grid_waves(uint64_t nrec) {
  uint64_t b = (nrec + WPB - 1) / WPB; /* one record per wave per iteration */
  if (b > 8192) b = 8192;
  if (b == 0) b = 1;
  return (uint32_t)b;
}

/* Pooled device allocator: size-class (next power of two) free lists.
 * hipMalloc/hipFree of multi-GB buffers costs hundreds of ms; the shuffle
 * engine's buffer sizes recur every spill/step, so pooling removes that
 * entirely (288 GB HBM makes holding the pool cheap). */
#include <unordered_map>
/* host-side pool/registry lock: the C-ABI contract allows the producer
   thread to differ from the flush thread; device work is still
   one stream, but the allocator bookkeeping must not race */
static std::mutex& pool_mu() {
  static std::mutex m;
  return m;
}
static std::unordered_map<size_t, std::vector<void*>>& pool_map() {
  static std::unordered_map<size_t, std::vector<void*>> m;
  return m;
}
/* telemetry (bytes): in-use = handed to callers; held = hipMalloc'd total;
 * peak tracks in-use; drops counts drop-the-pool retries.  Guarded by
 * pool_mu like the maps. */
static size_t g_pool_inuse = 0, g_pool_held = 0, g_pool_peak = 0;
static uint64_t g_pool_drops = 0;
static size_t pool_class(size_t n) {
  /* pow2 classes up to 1 GiB; 256 MiB steps beyond (a 105 GB stream must
     not round to 128 GB on a 288 GB device).  The class is computed on the
     UNPADDED size — the +256 B overread pad is added to the physical
     hipMalloc instead (pool_alloc), so exactly-pow2 buffers do not jump a
     class (jumping inflated the 1e9-record working set past the drop-pool
     retry threshold: C3 1e9 fell 53 -> 14 GB/s). */
  const size_t GB = 1ull << 30;
  if (n > GB) return (n + (256ull << 20) - 1) & ~((256ull << 20) - 1);
  size_t c = 1 << 16;
  while (c < n) c <<= 1;
  return c;
}
/* NOTE on oversubscription: hipMalloc on this stack silently overcommits
 * (301 GB "held" succeeded on a 288 GB device at C3 1e9) — harmless while
 * the TOUCHED working set stays under physical HBM (in-use peak 218 GB
 * there), but FIRST TOUCHES of fresh allocations beyond capacity fault at
 * page-migration speed (~2 GB/s): an extra 8 GB scratch buffer cost ~4 s.
 * Alloc-time eviction of idle buffers was tried and measured WORSE (24 GB/s
 * vs 53 — hipFree device-syncs inside the hot path); the controls that work
 * are (a) not allocating n-scaled scratch at the large-config peak (the lk0
 * gate) and (b) the telemetry below to catch it. */
static int pool_alloc(size_t n, void** out, size_t* cls_out) {
  size_t cls = pool_class(n);
  std::lock_guard<std::mutex> lk(pool_mu());
  auto& fl = pool_map()[cls];
  if (!fl.empty()) {
    *out = fl.back(); fl.pop_back(); *cls_out = cls;
    g_pool_inuse += cls;
    if (g_pool_inuse > g_pool_peak) g_pool_peak = g_pool_inuse;
    return 0;
  }
  /* +256 B: word/window-granular kernels (funnel loads, CRC trailer word,
     the emit gather window) may read a little past a logical end; padding
     the physical allocation keeps every such read in bounds without
     inflating the size class */
  if (hipMalloc(out, cls + 256) != hipSuccess) {
    /* under pressure: drop the whole pool and retry once */
    size_t freed = 0;
    for (auto& kv : pool_map())
      for (void* q : kv.second) { (void)hipFree(q); freed += kv.first; }
    pool_map().clear();
    g_pool_held -= freed > g_pool_held ? g_pool_held : freed;
    g_pool_drops++;
    fprintf(stderr,
            "[tzs pool] drop-and-retry #%llu: want %.2f GB, in-use %.2f GB, "
            "freed %.2f GB of idle pool\n",
            (unsigned long long)g_pool_drops, cls / 1e9, g_pool_inuse / 1e9,
            freed / 1e9);
    if (hipMalloc(out, cls + 256) != hipSuccess) {
      snprintf(g_err, sizeof(g_err), "hipMalloc(%zu) failed", cls);
      return -12;
    }
  }
  g_pool_held += cls;
  g_pool_inuse += cls;
  if (g_pool_inuse > g_pool_peak) g_pool_peak = g_pool_inuse;
  *cls_out = cls;
  return 0;
}
static void pool_free(void* p, size_t cls) {
  if (!p) return;
  std::lock_guard<std::mutex> lk(pool_mu());
  pool_map()[cls].push_back(p);
  g_pool_inuse -= cls > g_pool_inuse ? g_pool_inuse : cls;
}

/* raw pool allocations (generator buffers): class tracked in a registry so
 * tzs_free_device can return them to the pool */
static std::unordered_map<void*, size_t>& pool_registry() {
  static std::unordered_map<void*, size_t> r;
  return r;
}
static int pool_alloc_raw(size_t n, void** out) {
  size_t cls = 0;
  int rc = pool_alloc(n, out, &cls);
  if (rc == 0) {
    std::lock_guard<std::mutex> lk(pool_mu());
    pool_registry()[*out] = cls;
  }
  return rc;
}
/* generator-produced uniformity hints, keyed by the offsets pointer: lets
 * the zero-copy absorb skip its full-n uniformity scan (the producer KNOWS
 * the records are fixed-stride) */
struct UniformHint { uint32_t rec_u, klen_u; uint64_t nbytes; };
static std::unordered_map<void*, UniformHint>& uniform_hints() {
  static std::unordered_map<void*, UniformHint> m;
  return m;
}
static void pool_free_raw(void* p) {
  if (!p) return;
  size_t cls = 0;
  {
    std::lock_guard<std::mutex> lk(pool_mu());
    uniform_hints().erase(p);  /* a reused pointer must not inherit a hint */
    auto it = pool_registry().find(p);
    if (it == pool_registry().end()) {
      (void)hipFree(p);
      return;
    }
    cls = it->second;
    pool_registry().erase(it);
  }
  pool_free(p, cls);
}

struct DBuf {
  void* p = nullptr;
  size_t sz = 0;   /* size class */
  DBuf() = default;
  DBuf(const DBuf&) = delete;
  DBuf& operator=(const DBuf&) = delete;
  DBuf(DBuf&& o) noexcept : p(o.p), sz(o.sz) { o.p = nullptr; o.sz = 0; }
  DBuf& operator=(DBuf&& o) noexcept {
    if (this != &o) { release(); p = o.p; sz = o.sz; o.p = nullptr; o.sz = 0; }
    return *this;
  }
  ~DBuf() { release(); }
  int alloc(size_t n) {
    if (n <= sz) return 0;
    release();
    return pool_alloc(n, &p, &sz);
  }
  void release() { if (p) { pool_free(p, sz); p = nullptr; sz = 0; } }
};

/* exclusive scan of u64 array (device), returns total via last+add trick */
static int scan_u64(const uint64_t* d_in, uint64_t* d_out, uint32_t n, uint64_t* h_total) {
  if (n == 0) { if (h_total) *h_total = 0; return 0; }
  {
    /* single-pass decoupled-lookback path (one read + one write) */
    static int no_oss = -1;
    if (no_oss < 0) no_oss = getenv("TZS_NO_SCAN_LOOKBACK") ? 1 : 0;
    uint32_t nb_l = nblocks_for(n, SCAN_TILE);
    if (!no_oss && nb_l > 8) {
      static thread_local DBuf st, tick;
      if (st.alloc(8ull * nb_l) == 0 && tick.alloc(32) == 0) {
        HIP_CHECK(hipMemsetAsync(st.p, 0, 8ull * nb_l));
        HIP_CHECK(hipMemsetAsync(tick.p, 0, 32));
        hipLaunchKernelGGL(k_scan_lookback, dim3(nb_l), dim3(BLOCK), 0, 0,
                           d_in, d_out, n, (uint64_t*)st.p, (uint32_t*)tick.p,
                           (uint32_t*)tick.p + 1, (uint64_t*)tick.p + 2);
        uint64_t hh[3] = {0, 0, 0}; /* {ticket|error, pad, total} */
        HIP_CHECK(hipMemcpy(hh, tick.p, 24, hipMemcpyDeviceToHost));
        uint32_t err = (uint32_t)(hh[0] >> 32);
        if (!err) {
          if (h_total) *h_total = hh[2];
          return 0;
        }
        /* lookback timeout: in-place input is already destroyed — that
           should never happen; fail loudly rather than silently misscan */
        if (d_in == d_out) FAIL(-70, "scan lookback timeout (in-place)");
        /* out-of-place: fall through to the classic path */
      }
    }
  }
  /* read the last input BEFORE the scan: for in-place scans (d_in == d_out)
     it is overwritten with the exclusive prefix */
  uint64_t last_val = 0;
  if (h_total)
    HIP_CHECK(hipMemcpy(&last_val, d_in + (n - 1), 8, hipMemcpyDeviceToHost));
  uint32_t nb = nblocks_for(n, SCAN_TILE);
  static thread_local DBuf sums1, sums2, sums3;
  if (sums1.alloc(sizeof(uint64_t) * (nb + 1))) return -12;
  hipLaunchKernelGGL(k_scan_partials, dim3(nb), dim3(BLOCK), 0, 0, d_in, d_out,
                     (uint64_t*)sums1.p, n);
  if (nb > 1) {
    uint32_t nb2 = nblocks_for(nb, SCAN_TILE);
    if (sums2.alloc(sizeof(uint64_t) * (nb2 + 1))) return -12;
    hipLaunchKernelGGL(k_scan_partials, dim3(nb2), dim3(BLOCK), 0, 0,
                       (uint64_t*)sums1.p, (uint64_t*)sums1.p, (uint64_t*)sums2.p, nb);
    if (nb2 > 1) {
      uint32_t nb3 = nblocks_for(nb2, SCAN_TILE);
      if (nb3 > 1) { snprintf(g_err, sizeof(g_err), "scan too deep"); return -22; }
      if (sums3.alloc(sizeof(uint64_t) * 2)) return -12;
      hipLaunchKernelGGL(k_scan_partials, dim3(1), dim3(BLOCK), 0, 0,
                         (uint64_t*)sums2.p, (uint64_t*)sums2.p, (uint64_t*)sums3.p, nb2);
      hipLaunchKernelGGL(k_scan_add, dim3(nb2), dim3(BLOCK), 0, 0,
                         (uint64_t*)sums1.p, (uint64_t*)sums2.p, nb);
    }
    hipLaunchKernelGGL(k_scan_add, dim3(nb), dim3(BLOCK), 0, 0, d_out,
                       (uint64_t*)sums1.p, n);
  }
  if (h_total) {
    uint64_t last_off = 0;
    HIP_CHECK(hipMemcpy(&last_off, d_out + (n - 1), 8, hipMemcpyDeviceToHost));
    *h_total = last_off + last_val;
  }
  return 0;
}

/* host-side RecTable builder: segment 0 inlined, the rest uploaded as a
 * device SegDesc array (any k — no kernarg bound). */
struct HostRT {
  std::vector<SegDesc> segs;
  std::vector<uint32_t> base; /* [nspills+1] */
  DBuf d_segs, d_base;
  RecTable rt = {};
  void add(const void* data, const void* off, const void* klen,
           uint32_t rec_u, uint32_t klen_u, uint32_t n) {
    if (base.empty()) base.push_back(0);
    SegDesc sd;
    sd.data = (const uint8_t*)data;
    sd.off = (const uint64_t*)off;
    sd.klen = (const uint32_t*)klen;
    sd.rec_u = rec_u;
    sd.klen_u = klen_u;
    segs.push_back(sd);
    base.push_back(base.back() + n);
  }
  int finish(int key_type) {
    int ns = (int)segs.size();
    memset(&rt, 0, sizeof(rt));
    rt.nspills = ns;
    rt.key_type = key_type;
    if (ns == 0) return 0;
    rt.data0 = segs[0].data; rt.off0 = segs[0].off; rt.klen0 = segs[0].klen;
    rt.rec_u0 = segs[0].rec_u; rt.klen_u0 = segs[0].klen_u;
    rt.n0 = base[1] - base[0];
    if (ns > 1) {
      if (d_segs.alloc(sizeof(SegDesc) * ns)) return -12;
      if (d_base.alloc(4ull * (ns + 1))) return -12;
      HIP_CHECK(hipMemcpyAsync(d_segs.p, segs.data(), sizeof(SegDesc) * ns,
                               hipMemcpyHostToDevice));
      HIP_CHECK(hipMemcpyAsync(d_base.p, base.data(), 4ull * (ns + 1),
                               hipMemcpyHostToDevice));
      rt.segs = (const SegDesc*)d_segs.p;
      rt.base = (const uint32_t*)d_base.p;
    }
    return 0;
  }
  void reset() {
    segs.clear();
    base.clear();
    memset(&rt, 0, sizeof(rt));
  }
};

static thread_local int64_t g_scatter_ns = 0;
static thread_local int64_t g_scatter_launches = 0;
static thread_local int64_t g_scatter_elems = 0;

static thread_local DBuf g_rs_counts, g_rs_offsets, g_rs_totals, g_rs_bases,
    g_rs_tk64, g_rs_tk32, g_rs_ta0, g_rs_ta1;
static thread_local DBuf g_rs_tb64;
static void radix_release_temps() {
  g_rs_counts.release(); g_rs_offsets.release();
  g_rs_tk64.release(); g_rs_tk32.release(); g_rs_ta0.release(); g_rs_ta1.release();
  g_rs_tb64.release();
}

/* stable LSD radix over KeyT with payload arrays. Sorts in place (ping-pong,
 * result left in the primary arrays). */
/* dbuf_key/dbuf_a0/dbuf_a1 may be passed so an odd pass count SWAPS the
 * buffer handles with the ping-pong temps instead of copying ~1.2 GB back. */
template <typename KeyT>
static int radix_sort(KeyT* d_key, uint32_t* d_a0, uint32_t* d_a1, uint32_t n,
                      int nbytes_key, int first_byte = 0,
                      DBuf* dbuf_key = nullptr, DBuf* dbuf_a0 = nullptr,
                      DBuf* dbuf_a1 = nullptr, uint64_t* d_b64 = nullptr,
                      DBuf* dbuf_b64 = nullptr) {
  if (n <= 1) return 0;
  uint32_t nb = nblocks_for(n, TILE);
  DBuf& counts = g_rs_counts; DBuf& offsets = g_rs_offsets;
  DBuf& totals = g_rs_totals; DBuf& bases = g_rs_bases;
  DBuf& tk64 = g_rs_tk64; DBuf& tk32 = g_rs_tk32;
  DBuf& ta0 = g_rs_ta0; DBuf& ta1 = g_rs_ta1;
  if (counts.alloc(sizeof(uint32_t) * nb * RADIX)) return -12;
  if (offsets.alloc(sizeof(uint32_t) * nb * RADIX)) return -12;
  if (totals.alloc(sizeof(uint32_t) * RADIX)) return -12;
  if (bases.alloc(sizeof(uint32_t) * RADIX)) return -12;
  DBuf& tk = (sizeof(KeyT) == 8) ? tk64 : tk32;
  if (tk.alloc(sizeof(KeyT) * n)) return -12;
  if (ta0.alloc(sizeof(uint32_t) * n)) return -12;
  bool has_a1 = d_a1 != nullptr;
  if (has_a1 && ta1.alloc(sizeof(uint32_t) * n)) return -12;
  bool has_b64 = d_b64 != nullptr;
  DBuf& tb64 = g_rs_tb64;
  if (has_b64 && tb64.alloc(sizeof(uint64_t) * n)) return -12;

  hipEvent_t evs[16], eve[16];
  int nev = 0;
  KeyT* kin = d_key;
  KeyT* kout = (KeyT*)tk.p;
  uint32_t* a0in = d_a0;
  uint32_t* a0out = (uint32_t*)ta0.p;
  uint32_t* a1in = d_a1;
  uint32_t* a1out = (uint32_t*)ta1.p;
  uint64_t* b64in = d_b64;
  uint64_t* b64out = (uint64_t*)tb64.p;
  int passes = 0;
  bool pass_skip[16] = {false};

  /* onesweep path (any key width / payload shape, incl. the refinement
     seg-sorts): one global-histogram read for all passes, then a single
     lookback kernel per pass.  Falls back to the classic 3-kernel pass on a
     lookback timeout. */
  static int use_onesweep = -1;
  if (use_onesweep < 0) {
    const char* e = getenv("TZS_ONESWEEP");
    use_onesweep = (e && e[0] == '0') ? 0 : 1;
  }
  /* payload-carrying onesweep (a1/b64 staged through LDS) measured SLOWER
     than the classic 3-kernel path on C3's refinement seg-sorts (48.9 ->
     44.9 GB/s): the 50-57 KB LDS footprint halves resident blocks and each
     seg-sort pays a host histogram sync.  The dispatch below supports every
     shape; the gate keeps onesweep to the base (key + record-id) sort where
     it wins. */
  /* payload shapes ride onesweep too since the barrier-light rewrite:
     u64+two-u32 payloads at 1024 threads (130 KB LDS, 16 waves/CU);
     u32-key+u64 payload at 256 (41 KB, 12 waves/CU — the 1024 variant
     would need 162 KB).  Pre-rewrite this measured slower and was gated
     off. */
  /* payload shapes (the refinement's level-key and run-id sorts) can ride
     onesweep at 256-thread tiles (the 1024 a1 instantiation exceeds LDS at
     12 rounds); re-testable now that the histogram prefix is device-side
     (TZS_OS_PAYLOAD=1) */
  static int os_payload = -1;
  if (os_payload < 0) {
    const char* e = getenv("TZS_OS_PAYLOAD");
    os_payload = (e && e[0] == '1') ? 1 : 0;
  }
  bool os_shape_ok = (!has_a1 && !has_b64) ||
                     (os_payload && !(has_a1 && has_b64));
  if (use_onesweep && os_shape_ok && n >= 20000) {
    int npasses = nbytes_key - first_byte;
    /* bigger blocks scale the tile while KEEPING 16 waves/CU: 512 threads =
       4096-elem tile (2 blocks x 8 waves), 1024 = 8192-elem tile (1 block x
       16 waves, ~135 KB LDS) — digit runs grow to ~32 x 12 B = 384 B so
       partial-line scatter waste nearly vanishes (measured 289/293/301 GB/s
       for 256/512/1024 at C2; TZS_OS_BLK overrides) */
    static int os_blk = -1;
    if (os_blk < 0) {
      const char* e = getenv("TZS_OS_BLK");
      os_blk = e ? atoi(e) : 1024;
      if (os_blk != 256 && os_blk != 512 && os_blk != 1024) os_blk = 1024;
    }
    int eff_blk = (has_a1 || has_b64) ? 256 : os_blk;
    uint32_t os_tile = (uint32_t)TILE_ROUNDS * (uint32_t)eff_blk;
    uint32_t nb_os = nblocks_for(n, os_tile);
    static thread_local DBuf gh, gbases, st, tick;
    if (gh.alloc(4u * npasses * RADIX)) return -12;
    if (gbases.alloc(4u * npasses * RADIX)) return -12;
    if (st.alloc(4ull * nb_os * RADIX)) return -12;
    if (tick.alloc(16)) return -12;
    HIP_CHECK(hipMemsetAsync(tick.p, 0, 16));  /* err word cleared once */
    HIP_CHECK(hipMemsetAsync(gh.p, 0, 4u * npasses * RADIX));
    hipLaunchKernelGGL((k_global_hist_all<KeyT>), dim3(grid1d(n)), dim3(BLOCK),
                       (uint32_t)(4 * npasses * RADIX), 0, kin, n,
                       first_byte, npasses, (uint32_t*)gh.p);
    hipLaunchKernelGGL(k_os_bases, dim3(npasses), dim3(RADIX), 0, 0,
                       (const uint32_t*)gh.p, (uint32_t*)gbases.p);
    for (int b = first_byte; b < nbytes_key; b++) {
      int p = b - first_byte;
      HIP_CHECK(hipMemsetAsync(st.p, 0, 4ull * nb_os * RADIX));
      /* reset only the ticket word — the error word ([1]) accumulates across
         passes and is checked once after the loop */
      HIP_CHECK(hipMemsetAsync(tick.p, 0, 4));
      if (nev < 16) { (void)hipEventCreate(&evs[nev]); (void)hipEventCreate(&eve[nev]);
                      (void)hipEventRecord(evs[nev]); }
      const uint32_t* pbases = (const uint32_t*)((uint32_t*)gbases.p + p * RADIX);
      if (has_b64) {
        hipLaunchKernelGGL((k_onesweep_pass<KeyT, false, true, 256>), dim3(nb_os),
                           dim3(256), 0, 0, kin, kout, a0in, a0out, nullptr,
                           nullptr, b64in, b64out, n, b, pbases, (uint32_t*)st.p,
                           (uint32_t*)tick.p, (uint32_t*)tick.p + 1);
      } else if (has_a1) {
        hipLaunchKernelGGL((k_onesweep_pass<KeyT, true, false, 256>), dim3(nb_os),
                           dim3(256), 0, 0, kin, kout, a0in, a0out, a1in, a1out,
                           nullptr, nullptr, n, b, pbases, (uint32_t*)st.p,
                           (uint32_t*)tick.p, (uint32_t*)tick.p + 1);
      } else if (os_blk == 1024)
        hipLaunchKernelGGL((k_onesweep_pass<KeyT, false, false, 1024>), dim3(nb_os),
                           dim3(1024), 0, 0, kin, kout, a0in, a0out, nullptr, nullptr,
                           nullptr, nullptr, n, b, pbases, (uint32_t*)st.p,
                           (uint32_t*)tick.p, (uint32_t*)tick.p + 1);
      else if (os_blk == 512)
        hipLaunchKernelGGL((k_onesweep_pass<KeyT, false, false, 512>), dim3(nb_os),
                           dim3(512), 0, 0, kin, kout, a0in, a0out, nullptr, nullptr,
                           nullptr, nullptr, n, b, pbases, (uint32_t*)st.p,
                           (uint32_t*)tick.p, (uint32_t*)tick.p + 1);
      else
        hipLaunchKernelGGL((k_onesweep_pass<KeyT, false, false, 256>), dim3(nb_os),
                           dim3(256), 0, 0, kin, kout, a0in, a0out, nullptr, nullptr,
                           nullptr, nullptr, n, b, pbases, (uint32_t*)st.p,
                           (uint32_t*)tick.p, (uint32_t*)tick.p + 1);
      if (nev < 16) { (void)hipEventRecord(eve[nev]); nev++; }
      std::swap(kin, kout);
      std::swap(a0in, a0out);
      if (has_a1) std::swap(a1in, a1out);
      if (has_b64) std::swap(b64in, b64out);
      passes++;
    }
    {
      /* one deferred timeout check for all passes (a per-pass sync read cost
         ~0.1 ms each; a timeout is a should-never-happen protocol failure
         and poisons every later pass anyway) */
      uint32_t h_err = 0;
      HIP_CHECK(hipMemcpy(&h_err, (uint32_t*)tick.p + 1, 4, hipMemcpyDeviceToHost));
      if (h_err) {
        snprintf(g_err, sizeof(g_err), "onesweep lookback timeout");
        return -70;
      }
    }
    goto finish;
  }

  /* classic path: one upfront all-pass histogram detects single-digit
     passes (Zipf/short keys leave low level-key bytes constant or zero) so
     they skip entirely — one extra read of the keys + one sync buys up to
     nbytes_key-1 saved passes */
  if (n >= 100000 && nbytes_key - first_byte > 1) {
    int npasses = nbytes_key - first_byte;
    static thread_local DBuf gh2;
    if (gh2.alloc(4u * npasses * RADIX) == 0) {
      HIP_CHECK(hipMemsetAsync(gh2.p, 0, 4u * npasses * RADIX));
      hipLaunchKernelGGL((k_global_hist_all<KeyT>), dim3(grid1d(n)), dim3(BLOCK),
                         (uint32_t)(4 * npasses * RADIX), 0, kin, n,
                         first_byte, npasses, (uint32_t*)gh2.p);
      std::vector<uint32_t> h_cnt2(npasses * RADIX);
      HIP_CHECK(hipMemcpy(h_cnt2.data(), gh2.p, 4u * npasses * RADIX,
                          hipMemcpyDeviceToHost));
      for (int p = 0; p < npasses && p < 16; p++)
        for (int d = 0; d < RADIX; d++)
          if (h_cnt2[p * RADIX + d] == n) { pass_skip[p] = true; break; }
    }
  }
  /* classic path at 512-thread blocks: 4096-element tiles keep 8 waves/CU
     (1-2 resident blocks) while doubling digit runs; the u64-key + u64-
     payload variant would exceed the 160 KB LDS at 512 and stays at 256
     (it is never dispatched at runtime — b64 rides u32 seg keys). */
  {
    static int cl_blk = -1;
    if (cl_blk < 0) {
      const char* e = getenv("TZS_CL_BLK");
      cl_blk = e ? atoi(e) : 256;   /* 512 measured neutral on the
                                        refinement shapes (55.7 vs 56.0
                                        GB/s C3) — they are bound by the
                                        compact/gather pattern, not by
                                        digit-run length */
      if (cl_blk != 256 && cl_blk != 512) cl_blk = 256;
    }
    uint32_t cl_tile = (uint32_t)TILE_ROUNDS * (uint32_t)cl_blk;
    uint32_t nb_cl = nblocks_for(n, cl_tile);
    if (nb_cl > nb) nb_cl = nb;  /* counts/offsets were sized for nb */
    bool use512 = (cl_blk == 512);
    if (sizeof(KeyT) == 8 && has_b64) use512 = false;  /* LDS > 160 KB */
  for (int b = first_byte; b < nbytes_key; b++) {
    if (b - first_byte < 16 && pass_skip[b - first_byte]) continue;
    uint32_t nbl = use512 ? nb_cl : nb;
    if (use512)
      hipLaunchKernelGGL((k_radix_hist<KeyT, 512>), dim3(nbl), dim3(512), 0, 0,
                         kin, n, b, (uint32_t*)counts.p);
    else
      hipLaunchKernelGGL((k_radix_hist<KeyT>), dim3(nbl), dim3(BLOCK), 0, 0, kin, n, b,
                         (uint32_t*)counts.p);
    hipLaunchKernelGGL(k_radix_scan_blocks, dim3(RADIX), dim3(BLOCK), 0, 0,
                       (uint32_t*)counts.p, nbl, (uint32_t*)offsets.p, (uint32_t*)totals.p);
    hipLaunchKernelGGL(k_radix_scan_digits, dim3(1), dim3(RADIX), 0, 0,
                       (uint32_t*)totals.p, (uint32_t*)bases.p);
    if (nev < 16) { (void)hipEventCreate(&evs[nev]); (void)hipEventCreate(&eve[nev]);
                    (void)hipEventRecord(evs[nev]); }
    if (has_b64) {
      if constexpr (sizeof(KeyT) == 4) {
        if (use512) {
          hipLaunchKernelGGL((k_radix_scatter<KeyT, false, true, 512>), dim3(nbl),
                             dim3(512), 0, 0, kin, kout, a0in, a0out, nullptr,
                             nullptr, b64in, b64out, n, b, (uint32_t*)offsets.p,
                             (uint32_t*)bases.p);
        } else {
          hipLaunchKernelGGL((k_radix_scatter<KeyT, false, true>), dim3(nbl),
                             dim3(BLOCK), 0, 0, kin, kout, a0in, a0out, nullptr,
                             nullptr, b64in, b64out, n, b, (uint32_t*)offsets.p,
                             (uint32_t*)bases.p);
        }
      } else {
        hipLaunchKernelGGL((k_radix_scatter<KeyT, false, true>), dim3(nbl),
                           dim3(BLOCK), 0, 0, kin, kout, a0in, a0out, nullptr,
                           nullptr, b64in, b64out, n, b, (uint32_t*)offsets.p,
                           (uint32_t*)bases.p);
      }
    } else if (has_a1) {
      if (use512)
        hipLaunchKernelGGL((k_radix_scatter<KeyT, true, false, 512>), dim3(nbl),
                           dim3(512), 0, 0, kin, kout, a0in, a0out, a1in, a1out,
                           nullptr, nullptr, n, b, (uint32_t*)offsets.p,
                           (uint32_t*)bases.p);
      else
        hipLaunchKernelGGL((k_radix_scatter<KeyT, true>), dim3(nbl), dim3(BLOCK),
                           0, 0, kin, kout, a0in, a0out, a1in, a1out, nullptr,
                           nullptr, n, b, (uint32_t*)offsets.p, (uint32_t*)bases.p);
    } else {
      if (use512)
        hipLaunchKernelGGL((k_radix_scatter<KeyT, false, false, 512>), dim3(nbl),
                           dim3(512), 0, 0, kin, kout, a0in, a0out, nullptr,
                           nullptr, nullptr, nullptr, n, b, (uint32_t*)offsets.p,
                           (uint32_t*)bases.p);
      else
        hipLaunchKernelGGL((k_radix_scatter<KeyT, false>), dim3(nbl), dim3(BLOCK),
                           0, 0, kin, kout, a0in, a0out, nullptr, nullptr, nullptr,
                           nullptr, n, b, (uint32_t*)offsets.p, (uint32_t*)bases.p);
    }
    if (nev < 16) { (void)hipEventRecord(eve[nev]); nev++; }
    std::swap(kin, kout);
    std::swap(a0in, a0out);
    if (has_a1) std::swap(a1in, a1out);
    if (has_b64) std::swap(b64in, b64out);
    passes++;
  }
  }
finish:
  (void)hipDeviceSynchronize();
  for (int e = 0; e < nev; e++) {
    float ms = 0;
    (void)hipEventElapsedTime(&ms, evs[e], eve[e]);
    g_scatter_ns += (int64_t)(ms * 1e6);
    g_scatter_launches += 1;
    g_scatter_elems += n;
    (void)hipEventDestroy(evs[e]); (void)hipEventDestroy(eve[e]);
  }
  if (passes & 1) {
    if (dbuf_key && dbuf_a0 && (!has_a1 || dbuf_a1) && (!has_b64 || dbuf_b64)) {
      /* hand the temp buffers to the caller; keep the old primaries as temps */
      std::swap(*dbuf_key, tk);
      std::swap(*dbuf_a0, ta0);
      if (has_a1) std::swap(*dbuf_a1, ta1);
      if (has_b64) std::swap(*dbuf_b64, tb64);
    } else {
      HIP_CHECK(hipMemcpyAsync(d_key, kin, sizeof(KeyT) * n, hipMemcpyDeviceToDevice));
      HIP_CHECK(hipMemcpyAsync(d_a0, a0in, sizeof(uint32_t) * n, hipMemcpyDeviceToDevice));
      if (has_a1)
        HIP_CHECK(hipMemcpyAsync(d_a1, a1in, sizeof(uint32_t) * n,
                                 hipMemcpyDeviceToDevice));
      if (has_b64)
        HIP_CHECK(hipMemcpyAsync(d_b64, b64in, sizeof(uint64_t) * n,
                                 hipMemcpyDeviceToDevice));
    }
  }
  return 0;
}

static int pbits_for(int32_t P) {
  if (P <= 1) return 0;
  int b = 0;
  int32_t v = P - 1;
  while (v) { b++; v >>= 1; }
  return b;
}

struct SpillData {
  /* columnar record set (original order) */
  DBuf data;   /* serialized records */
  DBuf off;    /* u64 [n+1] */
  DBuf klen;   /* u32 [n] */
  int64_t n = 0;
  uint8_t rle = 0;
  uint32_t rec_u = 0, klen_u = 0;
  /* retained sorted state: skey[i] = masked composite of the i-th sorted
     record, skey2[i] = the next 8 comparator-source bytes (zero-padded)
     from byte lo_c0, sidxb[i] = its ORIGINAL local record index.  Lets
     flush merge already-sorted spills (128-bit merge path) instead of
     re-sorting the union. */
  DBuf skey;   /* u64 [n] */
  DBuf skey2;  /* u64 [n] */
  DBuf sidxb;  /* u32 [n] */
  int lo_c0 = -1;
  int sort_sb = 0;        /* composite mask width (bytes) */
  int sort_ser_mode = 0;  /* 1 = serialized-byte composite (TezBytes var-len) */
  uint8_t sorted_valid = 0;
  uint8_t no_stream = 0;  /* reduce-side segment: no per-spill IFile emitted */
  /* externally-owned columnar view (add_sorted_segment): caller keeps the
     buffers alive until flush/close; release() must not free them */
  const void* xdata = nullptr;
  const uint64_t* xoff = nullptr;
  const uint32_t* xklen = nullptr;
  void release() {
    data.release(); off.release(); klen.release(); ifile.release();
    skey.release(); skey2.release(); sidxb.release();
  }
  /* emitted IFile bytes + host index */
  DBuf ifile;
  int64_t ifile_len = 0;
  std::vector<tzs_index_record> index;
};

} // namespace

struct tzs_sorter {
  tzs_conf conf;
  int pbits;
  /* absorbed (unsorted) current buffer */
  DBuf cur_data, cur_off, cur_klen, cur_part;
  int64_t cur_n = 0;
  uint64_t cur_bytes = 0;
  uint32_t cur_rec_u = 0;   /* uniform record bytes (0 = not uniform) */
  uint32_t cur_klen_u = 0;
  bool cur_first_batch = true;
  bool have_explicit_parts = false;
  bool combined_parts_valid = false;
  /* combiner output of the last sort_and_emit (adopted by spill(): the
     reference's later merges read the COMBINED spill files, so the spill's
     stored record set must be the folded one) */
  DBuf combine_data, combine_off, combine_klen;
  uint32_t combine_n = 0;
  bool combine_applied = false;
  /* host-path staging */
  std::vector<uint8_t> host_data;
  std::vector<uint64_t> host_off;
  std::vector<uint32_t> host_klen;
  std::vector<int32_t> host_part;
  /* spills */
  std::vector<SpillData*> spills;
  /* final output */
  DBuf final_ifile;
  int64_t final_len = 0;
  std::vector<tzs_index_record> final_index;
  bool flushed = false;
  /* counters + times */
  tzs_counters ctr = {};
  tzs_times times = {};
  /* scratch kept across calls */
  DBuf skey, sidx, eq, same, sizes, scan, parts_sorted;
  DBuf skey_lo;                            /* merged lo keys (flush merge) */
  DBuf comp_ifile;                         /* TIF\1 compressed final stream */
  int64_t comp_len = 0;
  std::vector<tzs_index_record> comp_index;
  /* final-sort metadata for the exchange path */
  std::vector<uint64_t> final_rec_ranges;  /* [P+1] record index ranges */
  HostRT final_hrt;  /* multi-segment flush tables: owns the device SegDesc
                        arrays that final_rt may point into */
  RecTable final_rt = {};
  uint32_t final_n = 0;
  DBuf col_data, col_off, col_klen;        /* permuted columnar view */
};

extern "C" void tzs_conf_default(tzs_conf* c, int32_t num_partitions) {
  memset(c, 0, sizeof(*c));
  c->num_partitions = num_partitions;
  c->key_type = TZS_KEY_BYTES;
  c->value_type = TZS_KEY_BYTES;
  c->comparator = TZS_CMP_TEZBYTES;
  c->rle = -1;
  c->send_empty_partition_details = 1;
  c->io_sort_factor = 100;          /* TezRuntimeConfiguration.java:106 */
  c->final_merge_enabled = 1;
  c->sort_buffer_bytes = 100ll << 20; /* io.sort.mb=100, :117 */
  c->device = -1;
  c->world_size = 1;
  c->rank = 0;
  c->combiner = 0;
  c->min_spills_for_combine = 3; /* TEZ_RUNTIME_COMBINE_MIN_SPILLS, PipelinedSorter.java:244 */
  c->discard_spill_streams = 0;
}

extern "C" int tzs_sorter_create(const tzs_conf* conf, tzs_sorter** out) {
  if (!conf || conf->num_partitions <= 0) FAIL(-22, "bad conf");
  if (ensure_device_constants()) return -70;
  tzs_sorter* s = new tzs_sorter();
  s->conf = *conf;
  s->pbits = pbits_for(conf->num_partitions);
  *out = s;
  return 0;
}

extern "C" int tzs_sorter_write(tzs_sorter* s, const void* key, int32_t klen,
                                const void* val, int32_t vlen, int32_t partition) {
  /* KeyValuesWriter.write — host staging path.  Mirrors collect()
   * (PipelinedSorter.java:399-467): bytes are copied immediately. */
  if (s->host_off.empty()) s->host_off.push_back(0);
  const uint8_t* k = (const uint8_t*)key;
  const uint8_t* v = (const uint8_t*)val;
  s->host_data.insert(s->host_data.end(), k, k + klen);
  s->host_data.insert(s->host_data.end(), v, v + vlen);
  s->host_off.push_back(s->host_data.size());
  s->host_klen.push_back((uint32_t)klen);
  s->host_part.push_back(partition);
  /* counters are updated when the staging is absorbed (write_batch) */
  if ((int64_t)s->host_data.size() + 16ll * (int64_t)s->host_klen.size()
      > s->conf.sort_buffer_bytes)
    return tzs_sorter_spill(s);
  return 0;
}

static int absorb_host_staging(tzs_sorter* s) {
  if (s->host_klen.empty()) return 0;
  int64_t n = (int64_t)s->host_klen.size();
  static_assert(sizeof(uint64_t) == 8, "");
  DBuf dd, doff, dkl, dpart;
  if (dd.alloc(s->host_data.size() ? s->host_data.size() : 1)) return -12;
  if (doff.alloc(sizeof(uint64_t) * (n + 1))) return -12;
  if (dkl.alloc(sizeof(uint32_t) * n)) return -12;
  if (dpart.alloc(sizeof(int32_t) * n)) return -12;
  HIP_CHECK(hipMemcpy(dd.p, s->host_data.data(), s->host_data.size(), hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(doff.p, s->host_off.data(), sizeof(uint64_t) * (n + 1), hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(dkl.p, s->host_klen.data(), sizeof(uint32_t) * n, hipMemcpyHostToDevice));
  bool have_parts = true;
  for (auto p : s->host_part) if (p < 0) { have_parts = false; break; }
  HIP_CHECK(hipMemcpy(dpart.p, s->host_part.data(), sizeof(int32_t) * n, hipMemcpyHostToDevice));
  int rc = tzs_sorter_write_batch_device(s, dd.p, (const uint64_t*)doff.p,
                                         (const uint32_t*)dkl.p,
                                         have_parts ? (const int32_t*)dpart.p : nullptr, n);
  /* write_batch copies, so free staging */
  dd.release(); doff.release(); dkl.release(); dpart.release();
  s->host_data.clear(); s->host_off.clear(); s->host_klen.clear(); s->host_part.clear();
  return rc;
}

#include <chrono>
extern "C" int tzs_sorter_write_batch_device(tzs_sorter* s, const void* d_data,
                                             const uint64_t* d_off, const uint32_t* d_klen,
                                             const int32_t* d_part, int64_t n) {
  auto t0 = std::chrono::steady_clock::now();
  if (n == 0) return 0;
  if (n > 4000000000ll) FAIL(-22, "batch too large (u32 record ids)");
  /* append into current buffer (device-side copy) */
  uint64_t nbytes = 0, first = 0;
  HIP_CHECK(hipMemcpy(&first, d_off, 8, hipMemcpyDeviceToHost));
  HIP_CHECK(hipMemcpy(&nbytes, d_off + n, 8, hipMemcpyDeviceToHost));
  nbytes -= first;
  if (first != 0) FAIL(-22, "d_off must start at 0");

  uint64_t old_bytes = s->cur_bytes;
  int64_t old_n = s->cur_n;
  /* grow buffers (simple realloc-copy; optimize later with chunked arenas) */
  DBuf nd, noff, nkl, npart;
  if (nd.alloc(old_bytes + nbytes ? old_bytes + nbytes : 1)) return -12;
  if (noff.alloc(sizeof(uint64_t) * (old_n + n + 1))) return -12;
  if (nkl.alloc(sizeof(uint32_t) * (old_n + n))) return -12;
  if (npart.alloc(sizeof(int32_t) * (old_n + n))) return -12;
  if (old_n) {
    HIP_CHECK(hipMemcpyAsync(nd.p, s->cur_data.p, old_bytes, hipMemcpyDeviceToDevice));
    HIP_CHECK(hipMemcpyAsync(noff.p, s->cur_off.p, sizeof(uint64_t) * old_n, hipMemcpyDeviceToDevice));
    HIP_CHECK(hipMemcpyAsync(nkl.p, s->cur_klen.p, sizeof(uint32_t) * old_n, hipMemcpyDeviceToDevice));
    HIP_CHECK(hipMemcpyAsync(npart.p, s->cur_part.p, sizeof(int32_t) * old_n, hipMemcpyDeviceToDevice));
  }
  HIP_CHECK(hipMemcpyAsync((uint8_t*)nd.p + old_bytes, d_data, nbytes, hipMemcpyDeviceToDevice));
  HIP_CHECK(hipMemcpyAsync((uint32_t*)nkl.p + old_n, d_klen, sizeof(uint32_t) * n, hipMemcpyDeviceToDevice));
  /* offsets: shift by old_bytes */
  hipLaunchKernelGGL(k_shift_offsets, dim3(grid1d(n + 1)), dim3(BLOCK), 0, 0,
                     d_off, (uint64_t*)noff.p + old_n, old_bytes, n + 1);
  std::swap(s->cur_data, nd); nd.release();
  std::swap(s->cur_off, noff); noff.release();
  std::swap(s->cur_klen, nkl); nkl.release();
  /* partitions: explicit ones are stored; HashPartitioner placement is
     computed fused into the composite build at sort time */
  if (d_part) {
    HIP_CHECK(hipMemcpyAsync((int32_t*)npart.p + old_n, d_part, sizeof(int32_t) * n,
                             hipMemcpyDeviceToDevice));
    s->have_explicit_parts = true;
  } else if (s->have_explicit_parts) {
    FAIL(-22, "cannot mix explicit and computed partitions in one sorter");
  }
  std::swap(s->cur_part, npart); npart.release();
  {
    static thread_local DBuf mm;
    if (mm.alloc(32)) return -12;
    uint64_t init[4] = {~0ull, 0, ~0ull, 0};
    HIP_CHECK(hipMemcpyAsync(mm.p, init, 32, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(k_check_uniform, dim3(grid1d(n)), dim3(BLOCK), 0, 0,
                       d_off, d_klen, n, (uint64_t*)mm.p);
    uint64_t res[4];
    HIP_CHECK(hipMemcpy(res, mm.p, 32, hipMemcpyDeviceToHost));
    bool uni = (res[0] == res[1]) && (res[2] == res[3]) && res[0] <= 0xFFFFFFFFull;
    if (getenv("TZS_NO_UNIFORM")) uni = false;
    if (s->cur_first_batch) {
      s->cur_rec_u = uni ? (uint32_t)res[0] : 0;
      s->cur_klen_u = uni ? (uint32_t)res[2] : 0;
      s->cur_first_batch = false;
    } else if (!uni || s->cur_rec_u != (uint32_t)res[0] ||
               s->cur_klen_u != (uint32_t)res[2]) {
      s->cur_rec_u = 0;
      s->cur_klen_u = 0;
    }
  }
  s->cur_n = old_n + n;
  s->cur_bytes = old_bytes + nbytes;
  s->ctr.output_records += n;
  s->ctr.output_bytes += (int64_t)nbytes; /* serialized k+v bytes (framing incl.) */
  HIP_CHECK(hipDeviceSynchronize());
  s->times.absorb_ns += std::chrono::duration_cast<std::chrono::nanoseconds>(
      std::chrono::steady_clock::now() - t0).count();
  return 0;
}

extern "C" int tzs_sorter_write_batch_device_adopt(tzs_sorter* s, void* d_data,
                                                   uint64_t* d_off, uint32_t* d_klen,
                                                   int32_t* d_part, int64_t n) {
  /* Zero-copy absorb: the sorter takes OWNERSHIP of device buffers that were
   * allocated through this library (tzs_malloc_device / tzs_generate).  The
   * reference's collect() copies because its input arrives record-at-a-time
   * from the processor (PipelinedSorter.java:399-467); a device-resident
   * producer handing over whole buffers is the MI355X-native equivalent and
   * skips the 2x-payload D2D copy.  Only valid as the FIRST batch of a spill
   * and only for registry-backed pointers; the caller must not touch or free
   * the buffers afterwards. */
  auto t0 = std::chrono::steady_clock::now();
  if (n == 0) return 0;
  if (n > 4000000000ll) FAIL(-22, "batch too large (u32 record ids)");
  if (s->cur_n != 0 || !s->host_klen.empty())
    FAIL(-22, "adopt requires an empty current buffer (first batch of spill)");
  std::unique_lock<std::mutex> reg_lk(pool_mu());
  auto& reg = pool_registry();
  auto id = reg.find(d_data), io = reg.find(d_off), ik = reg.find(d_klen);
  auto ip = d_part ? reg.find(d_part) : reg.end();
  if (id == reg.end() || io == reg.end() || ik == reg.end() ||
      (d_part && ip == reg.end()))
    FAIL(-22, "adopt requires buffers allocated by tzs_malloc_device/tzs_generate");
  size_t cls_d = id->second, cls_o = io->second, cls_k = ik->second;
  size_t cls_p = d_part ? ip->second : 0;
  reg_lk.unlock();
  uint64_t nbytes = 0;
  bool hinted = false;
  {
    std::lock_guard<std::mutex> lk(pool_mu());
    auto hh = uniform_hints().find(d_off);
    if (hh != uniform_hints().end() && !getenv("TZS_NO_UNIFORM")) {
      s->cur_rec_u = hh->second.rec_u;
      s->cur_klen_u = hh->second.klen_u;
      nbytes = hh->second.nbytes;
      s->cur_first_batch = false;
      uniform_hints().erase(hh);
      hinted = true;
    }
  }
  if (!hinted) {
    uint64_t first = 0;
    HIP_CHECK(hipMemcpy(&first, d_off, 8, hipMemcpyDeviceToHost));
    HIP_CHECK(hipMemcpy(&nbytes, d_off + n, 8, hipMemcpyDeviceToHost));
    if (first != 0) FAIL(-22, "d_off must start at 0");
    /* uniform-record detection (same rule as the copy path) */
    static thread_local DBuf mm;
    if (mm.alloc(32)) return -12;
    uint64_t init[4] = {~0ull, 0, ~0ull, 0};
    HIP_CHECK(hipMemcpyAsync(mm.p, init, 32, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(k_check_uniform, dim3(grid1d(n)), dim3(BLOCK), 0, 0,
                       d_off, d_klen, n, (uint64_t*)mm.p);
    uint64_t res[4];
    HIP_CHECK(hipMemcpy(res, mm.p, 32, hipMemcpyDeviceToHost));
    bool uni = (res[0] == res[1]) && (res[2] == res[3]) && res[0] <= 0xFFFFFFFFull;
    if (getenv("TZS_NO_UNIFORM")) uni = false;
    s->cur_rec_u = uni ? (uint32_t)res[0] : 0;
    s->cur_klen_u = uni ? (uint32_t)res[2] : 0;
    s->cur_first_batch = false;
  }
  /* take ownership: registry entry -> DBuf (released to the pool later) */
  {
    std::lock_guard<std::mutex> lk(pool_mu());
    reg.erase(d_data);
    reg.erase(d_off);
    reg.erase(d_klen);
    if (d_part) reg.erase(d_part);
  }
  s->cur_data.release(); s->cur_data.p = d_data; s->cur_data.sz = cls_d;
  s->cur_off.release();  s->cur_off.p = d_off;   s->cur_off.sz = cls_o;
  s->cur_klen.release(); s->cur_klen.p = d_klen; s->cur_klen.sz = cls_k;
  if (d_part) {
    s->cur_part.release(); s->cur_part.p = d_part; s->cur_part.sz = cls_p;
    s->have_explicit_parts = true;
  } else {
    if (s->cur_part.alloc(sizeof(int32_t) * n)) return -12;
    s->have_explicit_parts = false;
  }
  s->cur_n = n;
  s->cur_bytes = nbytes;
  s->ctr.output_records += n;
  s->ctr.output_bytes += (int64_t)nbytes;
  HIP_CHECK(hipDeviceSynchronize());
  s->times.absorb_ns += std::chrono::duration_cast<std::chrono::nanoseconds>(
      std::chrono::steady_clock::now() - t0).count();
  return 0;
}

__global__ void k_shift_offsets(const uint64_t* src, uint64_t* dst, uint64_t shift,
                                int64_t n) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    dst[i] = src[i] + shift;
}

/* composite-parameter derivation shared by the sort and merge paths */
struct SortParams {
  int ser_mode;
  int SB;
  int ref_pb;
  int proxy_w;
};
static SortParams derive_sort_params(tzs_sorter* s,
                                     const std::vector<SegDesc>& hsegs,
                                     uint32_t n) {
  SortParams sp;
  int P = s->conf.num_partitions;
  int pbits = s->pbits;
  /* TezBytes order is (partition, truncated proxy, serialized bytes).  When
     every record's serialized klen is equal (uniform spills), the length
     word is constant and the truncated proxy is a prefix function of the
     content, so the order reduces to (partition, content) — use the cheap
     content composite (full 58-bit discrimination) in that case; the
     faithful serialized composite only for variable-length keys. */
  sp.ser_mode = (s->conf.comparator == TZS_CMP_TEZBYTES) ? 1 : 0;
  if (sp.ser_mode) {
    bool uniform_klen = !hsegs.empty();
    uint32_t k0 = hsegs.empty() ? 0 : hsegs[0].klen_u;
    for (auto& sd : hsegs)
      if (sd.klen_u == 0 || sd.klen_u != k0) { uniform_klen = false; break; }
    if (uniform_klen && k0 != 0) sp.ser_mode = 0;
  }
  sp.ref_pb = 1; /* bitcount(P)+1, PipelinedSorter.java:165 */
  for (int v2 = P; v2; v2 >>= 1) sp.ref_pb++;
  sp.proxy_w = 24 - sp.ref_pb;
  if (sp.proxy_w < 0) sp.proxy_w = 0;
  /* adaptive radix width: enough composite bits that expected tie-involved
     records stay below ~1% of n (ties go through refinement anyway):
     bits = pbits + log2(n) + 6.  Tests at small n exercise refinement hard. */
  int needed_bits = pbits + 6;
  for (uint64_t v = n; v; v >>= 1) needed_bits++;
  int SB = (needed_bits + 7) / 8;
  if (SB < 2) SB = 2;
  if (sp.ser_mode) {
    /* partition + surviving proxy bits must lie inside the sorted bytes */
    int minsb = (pbits + sp.proxy_w + 7) / 8;
    if (SB < minsb) SB = minsb;
  }
  if (s->conf.comparator == TZS_CMP_TEXT) {
    /* natural-language keys share long prefixes (Zipf words): ties explode
       the refinement stage (profiled ~40% of C3's sort time) — two extra
       radix passes are far cheaper */
    SB = 8;
  }
  if (SB > 8) SB = 8;
  sp.SB = SB;
  return sp;
}

static int refine_and_emit(tzs_sorter* s, std::vector<SegDesc> hsegs,
                           std::vector<uint32_t> hbase,
                           RecTable rt, uint32_t n, int SB, int ser_mode,
                           const uint8_t* h_spill_rle, int nspills_rle,
                           SpillData* outsp, bool apply_combine,
                           SpillData* retain, const uint64_t* d_lo);

/* ---- the core: sort current buffer + emit IFile segments ---- */
static int sort_and_emit(tzs_sorter* s, HostRT& hrt, uint32_t n,
                         const int32_t* d_part_unsorted,
                         const uint8_t* h_spill_rle, int nspills_rle,
                         SpillData* outsp, bool apply_combine = false,
                         SpillData* retain = nullptr) {
  tzs_times& T = s->times;
  g_scatter_ns = 0; g_scatter_launches = 0; g_scatter_elems = 0;
  hipEvent_t ev[4];
  for (auto& e : ev) (void)hipEventCreate(&e);
  (void)hipEventRecord(ev[0]);

  RecTable rt = hrt.rt;
  int P = s->conf.num_partitions;
  int pbits = s->pbits;
  SortParams prm = derive_sort_params(s, hrt.segs, n);
  int ser_mode = prm.ser_mode, SB = prm.SB;
  /* 1. composites */
  if (s->skey.alloc(sizeof(uint64_t) * n)) return -12;
  if (s->sidx.alloc(sizeof(uint32_t) * n)) return -12;
  uint64_t* d_key = (uint64_t*)s->skey.p;
  uint32_t* d_idx = (uint32_t*)s->sidx.p;
  hipLaunchKernelGGL(k_build_composite2, dim3(grid1d(n)), dim3(BLOCK), 0, 0, rt,
                     d_part_unsorted, P, pbits, prm.ref_pb, SB, ser_mode, d_key,
                     d_idx, n);
  (void)hipEventRecord(ev[1]);

  /* 2. base radix sort over the top SB bytes of the composite */
  int rc = radix_sort<uint64_t>(d_key, d_idx, nullptr, n, 8, 8 - SB,
                                &s->skey, &s->sidx, nullptr);
  if (rc) return rc;
  (void)hipEventRecord(ev[2]);
  (void)hipEventSynchronize(ev[2]);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, ev[0], ev[1]); T.composite_ns += (int64_t)(ms * 1e6);
  (void)hipEventElapsedTime(&ms, ev[1], ev[2]); T.sort_ns += (int64_t)(ms * 1e6);
  (void)hipEventElapsedTime(&ms, ev[0], ev[2]); T.total_ns += (int64_t)(ms * 1e6);
  for (auto& e : ev) (void)hipEventDestroy(e);
  return refine_and_emit(s, hrt.segs, hrt.base, rt, n, SB, ser_mode,
                         h_spill_rle, nspills_rle, outsp, apply_combine,
                         retain, nullptr);
}

/* refinement + combiner + IFile emit + CRC over an already-ordered view:
 * s->skey = masked composites in final base-sort order, s->sidx = global
 * record ids.  Entered from sort_and_emit (base radix sort) or from the
 * flush merge path (k_merge_path tree over retained spill orders). */
static int refine_and_emit(tzs_sorter* s, std::vector<SegDesc> hsegs,
                           std::vector<uint32_t> hbase,
                           RecTable rt, uint32_t n, int SB, int ser_mode,
                           const uint8_t* h_spill_rle, int nspills_rle,
                           SpillData* outsp, bool apply_combine,
                           SpillData* retain, const uint64_t* d_lo) {
  tzs_times& T = s->times;
  hipEvent_t ev[10];
  for (auto& e : ev) (void)hipEventCreate(&e);
  int P = s->conf.num_partitions;
  int pbits = s->pbits;
  int ref_pb = 1;
  for (int v2 = P; v2; v2 >>= 1) ref_pb++;
  int proxy_w = 24 - ref_pb;
  if (proxy_w < 0) proxy_w = 0;
  uint64_t* d_key = (uint64_t*)s->skey.p;
  uint32_t* d_idx = (uint32_t*)s->sidx.p;
  int rc = 0;
  (void)hipEventRecord(ev[2]);

  /* 3. refinement levels */
  if (s->eq.alloc(n)) return -12;
  if (s->parts_sorted.alloc(sizeof(uint32_t) * n)) return -12;
  uint8_t* d_eq = (uint8_t*)s->eq.p;
  if (d_lo)
    hipLaunchKernelGGL(k_eq_init2, dim3(grid1d(n)), dim3(BLOCK), 0, 0, d_key,
                       d_lo, d_eq, pbits, (uint32_t*)s->parts_sorted.p, n);
  else
    hipLaunchKernelGGL(k_eq_init, dim3(grid1d(n)), dim3(BLOCK), 0, 0, d_key, d_eq,
                       pbits, (uint32_t*)s->parts_sorted.p, n);
  /* max content length: for refinement level count */
  /* refinement start byte within the comparator's byte source: sorted-covered
     bits are 8*SB - pbits (content mode) or 8*SB - pbits - 24 past the proxy
     (serialized mode; the proxy is rechecked implicitly because serialized
     bytes repeat the content after the 4B length) */
  int c0 = ser_mode ? (8 * SB - pbits - proxy_w) / 8 : (8 * SB - pbits) / 8;
  if (c0 < 0) c0 = 0;
  /* the merged-lo entry path already compared 8 more source bytes */
  int c0_eff = d_lo ? c0 + 8 : c0;
  /* determine max clen lazily: use a safe cap by scanning klen on host?  We
     compute it from the conf: key_type BYTES => clen = klen-4 (max over
     spills).  For TEXT, clen <= klen-1.  Host keeps max_klen per spill. */
  uint32_t max_klen = 0;
  {
    /* reduce over d_klen of every spill in rt — small kernel; here use host
       copy of spill maxima maintained at absorb time.  For simplicity round 1:
       copy klen array and reduce on host only once per spill (n small) is too
       slow for 1e8; do a device reduction. */
    bool need_dev = false;
    for (size_t sp = 0; sp < hsegs.size(); sp++) {
      if (hsegs[sp].klen_u) {
        if (hsegs[sp].klen_u > max_klen) max_klen = hsegs[sp].klen_u;
      } else {
        need_dev = true;
      }
    }
    if (need_dev) {
      static thread_local DBuf dmax;
      if (dmax.alloc(4)) return -12;
      HIP_CHECK(hipMemsetAsync(dmax.p, 0, 4));
      for (size_t sp = 0; sp < hsegs.size(); sp++) {
        if (hsegs[sp].klen_u) continue;
        uint32_t cnt = hbase[sp + 1] - hbase[sp];
        if (!cnt) continue;
        hipLaunchKernelGGL(k_max_u32, dim3(grid1d(cnt)), dim3(BLOCK), 0, 0,
                           hsegs[sp].klen, cnt, (uint32_t*)dmax.p);
      }
      uint32_t dm = 0;
      HIP_CHECK(hipMemcpy(&dm, dmax.p, 4, hipMemcpyDeviceToHost));
      if (dm > max_klen) max_klen = dm;
    }
  }
  int max_clen = (int)max_klen; /* upper bound on content length */
  static thread_local DBuf eqcnt;
  static thread_local DBuf lkey, seg, pos, slotpos;
  static thread_local DBuf lk0;
  if (eqcnt.alloc(4)) return -12;

  bool will_combine = apply_combine && s->conf.combiner == 1;
  bool lk0_retain_ready = false; /* dense level-0 keys double as the lo
                                    retention source (same byte range) */
  int level = 1;
  const int max_levels = (max_clen > c0_eff) ? (max_clen - c0_eff + 7) / 8 : 0;
  /* initial ambiguity count; later levels get it from k_eq_update's
     survivor counter (saves a full-n read per level) */
  uint32_t neq = 0;
  {
    HIP_CHECK(hipMemsetAsync(eqcnt.p, 0, 4));
    hipLaunchKernelGGL(k_count_nonzero_u8, dim3(grid1d(n)), dim3(BLOCK), 0, 0, d_eq, n,
                       (uint32_t*)eqcnt.p);
    HIP_CHECK(hipMemcpy(&neq, eqcnt.p, 4, hipMemcpyDeviceToHost));
  }
  for (int li = 0; li <= max_levels; li++) {
    if (neq == 0) break;
    int use_len = (li == max_levels); /* final tiebreak: content length */
    /* m (in-run elements) <= 2*neq: a run of r equal elements carries r-1
       eq flags and r in-run members, r <= 2(r-1) for r >= 2 */
    uint64_t mcap64 = 2ull * neq;
    uint32_t mcap = (uint32_t)(mcap64 < n ? mcap64 : n);
    if (lkey.alloc(sizeof(uint64_t) * mcap)) return -12;
    if (seg.alloc(sizeof(uint32_t) * mcap)) return -12;
    if (pos.alloc(sizeof(uint32_t) * mcap)) return -12;
    int lb0 = c0_eff + 8 * li;
    uint64_t ptotal = 0;
    {
      /* dense original-order level-key build when most records are still
         ambiguous: coalesced reads beat the random gather.  Gated to
         n <= 3e8: the 8n-byte lk0 buffer at a 1e9-record merge pushed the
         peak working set into pool-drop/hipMalloc churn. */
      const uint64_t* lk0p = nullptr;
      /* !lk0_retain_ready: a later level must not overwrite the retained
         level-0 content (it gathers through sidx after the loop) */
      if (3ull * neq >= n && n <= 300000000u && !lk0_retain_ready) {
        if (lk0.alloc(8ull * n)) return -12;
        hipLaunchKernelGGL(k_build_lkeys, dim3(grid1d(n)), dim3(BLOCK), 0, 0, rt,
                           lb0, use_len, ser_mode, (uint64_t*)lk0.p, n);
        lk0p = (const uint64_t*)lk0.p;
      }
      /* fused flags + lookback scan + compaction */
      static thread_local DBuf rc_st, rc_tick;
      uint32_t nb_rc = nblocks_for(n, SCAN_TILE);
      if (rc_st.alloc(8ull * nb_rc)) return -12;
      if (rc_tick.alloc(32)) return -12;
      HIP_CHECK(hipMemsetAsync(rc_st.p, 0, 8ull * nb_rc));
      HIP_CHECK(hipMemsetAsync(rc_tick.p, 0, 32));
      hipLaunchKernelGGL(k_refine_compact_lb, dim3(nb_rc), dim3(BLOCK), 0, 0, rt,
                         d_idx, d_eq, n, lb0, use_len, ser_mode, lk0p,
                         (uint64_t*)lkey.p, (uint32_t*)seg.p, (uint32_t*)pos.p,
                         (uint64_t*)rc_st.p, (uint32_t*)rc_tick.p,
                         (uint32_t*)rc_tick.p + 1, (uint64_t*)rc_tick.p + 2);
      uint64_t hh[3] = {0, 0, 0};
      HIP_CHECK(hipMemcpy(hh, rc_tick.p, 24, hipMemcpyDeviceToHost));
      if ((uint32_t)(hh[0] >> 32)) FAIL(-70, "refine compact lookback timeout");
      ptotal = hh[2];
      if (li == 0 && !use_len && lk0p && retain && !will_combine)
        lk0_retain_ready = true; /* keep the buffer for retention below */
      else if (!lk0_retain_ready)
        lk0.release();
    }
    uint64_t nruns = ptotal & 0xFFFFFFFFu;
    uint32_t m = (uint32_t)(ptotal >> 32);
    if (m == 0) break;
    if (slotpos.alloc(sizeof(uint32_t) * m)) return -12;
    /* slotpos = copy of pos (ascending) before sort */
    HIP_CHECK(hipMemcpyAsync(slotpos.p, pos.p, sizeof(uint32_t) * m,
                             hipMemcpyDeviceToDevice));
    /* sort compact by (seg, lkey): LSD lkey bytes then seg bytes; the seg
       passes carry lkey as a 64-bit payload so it stays aligned for the
       eq update (re-gathering it measured ~210 ms/step at C3 1e9) */
    int segbytes = 1;
    while ((nruns >> (8 * segbytes)) != 0 && segbytes < 4) segbytes++;
    rc = radix_sort<uint64_t>((uint64_t*)lkey.p, (uint32_t*)seg.p, (uint32_t*)pos.p, m, 8);
    if (rc) return rc;
    rc = radix_sort<uint32_t>((uint32_t*)seg.p, (uint32_t*)pos.p, nullptr, m, segbytes,
                              0, nullptr, nullptr, nullptr, (uint64_t*)lkey.p, &lkey);
    if (rc) return rc;
    /* apply permutation to d_idx in place ({pos} == {slotpos} as position
       sets, so only the m in-run slots change) and update eq */
    static thread_local DBuf idx_tmp;
    if (idx_tmp.alloc(sizeof(uint32_t) * m)) return -12;
    hipLaunchKernelGGL(k_refine_gather, dim3(grid1d(m)), dim3(BLOCK), 0, 0,
                       (const uint32_t*)pos.p, d_idx, (uint32_t*)idx_tmp.p, m);
    hipLaunchKernelGGL(k_refine_scatter, dim3(grid1d(m)), dim3(BLOCK), 0, 0,
                       (const uint32_t*)slotpos.p, (const uint32_t*)idx_tmp.p,
                       d_idx, m);
    HIP_CHECK(hipMemsetAsync(eqcnt.p, 0, 4));
    hipLaunchKernelGGL(k_eq_update, dim3(grid1d(m)), dim3(BLOCK), 0, 0,
                       (const uint64_t*)lkey.p, (const uint32_t*)seg.p,
                       (const uint32_t*)slotpos.p, d_eq, m, (uint32_t*)eqcnt.p);
    HIP_CHECK(hipMemcpy(&neq, eqcnt.p, 4, hipMemcpyDeviceToHost));
    level++;
  }
  (void)hipEventRecord(ev[3]);
  /* large-n headroom: refinement scratch and radix ping-pong temps are dead
     from here; return them to the pool before the output-stream allocation
     (at C3's 1e9 records these hold ~60 GB) */
  lkey.release(); seg.release(); pos.release(); slotpos.release();
  radix_release_temps();

  /* retain the refined sorted order for the flush merge path (VERDICT r1
     #1): composites move out (nothing below reads them), sorted ids are
     copied (s->sidx stays live for emit + sorted_columnar).  Combiner
     spills retain the FOLDED set instead (built in the combiner branch). */
  if (retain && !will_combine) {
    std::swap(retain->skey, s->skey);
    if (retain->sidxb.alloc(4ull * n)) return -12;
    HIP_CHECK(hipMemcpyAsync(retain->sidxb.p, s->sidx.p, 4ull * n,
                             hipMemcpyDeviceToDevice));
    /* lo keys: comparator-source bytes [c0, c0+8) per sorted record — the
       refinement's dense level-0 build is the same array when it ran;
       otherwise build it now, then one 8B gather into sorted order */
    if (retain->skey2.alloc(8ull * n)) return -12;
    {
      if (!lk0_retain_ready) {
        if (lk0.alloc(8ull * n)) return -12;
        hipLaunchKernelGGL(k_build_lkeys, dim3(grid1d(n)), dim3(BLOCK), 0, 0, rt,
                           c0, 0, ser_mode, (uint64_t*)lk0.p, n);
      }
      hipLaunchKernelGGL(k_gather_merge_hi, dim3(grid1d(n)), dim3(BLOCK), 0, 0,
                         (const uint64_t*)lk0.p, (const uint32_t*)s->sidx.p,
                         (uint64_t*)retain->skey2.p, 0ull, n);
      lk0.release();
    }
    retain->lo_c0 = c0;
    retain->sort_sb = SB;
    retain->sort_ser_mode = ser_mode;
    retain->sorted_valid = 1;
  }

  /* 4. writer-rle decision + same flags (neq carried out of the refinement
     loop — after the final level eq[i] means full-key-equal) */
  uint32_t neq_final = neq;
  int writer_rle;
  if (s->conf.rle >= 0) {
    writer_rle = s->conf.rle;
  } else if (nspills_rle > 1) {
    /* multi-spill merge: the auto gate counts SAME_KEY machine events —
       equal keys meeting ACROSS segments or carried by an RLE'd source
       stream — exactly TezMerger's emergent rle trace (the oracle's
       merge gate; a raw adjacent-equal count would over-RLE unions whose
       within-spill duplicates sat below each spill's own gate).  Computed
       by running the provenance kernel with writer_rle=0 and counting. */
    if (s->same.alloc(n)) return -12;
    {
      static thread_local DBuf d_sprle0, cnt0;
      if (d_sprle0.alloc(nspills_rle)) return -12;
      if (cnt0.alloc(4)) return -12;
      HIP_CHECK(hipMemcpyAsync(d_sprle0.p, h_spill_rle, nspills_rle,
                               hipMemcpyHostToDevice));
      hipLaunchKernelGGL(k_writer_same, dim3(grid1d(n)), dim3(BLOCK), 0, 0, rt,
                         d_idx, d_eq, 0, (const uint8_t*)d_sprle0.p,
                         (uint8_t*)s->same.p, n);
      HIP_CHECK(hipMemsetAsync(cnt0.p, 0, 4));
      hipLaunchKernelGGL(k_count_nonzero_u8, dim3(grid1d(n)), dim3(BLOCK), 0, 0,
                         (const uint8_t*)s->same.p, n, (uint32_t*)cnt0.p);
      uint32_t ev = 0;
      HIP_CHECK(hipMemcpy(&ev, cnt0.p, 4, hipMemcpyDeviceToHost));
      writer_rle = ((uint64_t)ev * 10 > n) ? 1 : 0;
    }
  } else {
    /* single spill: adjacent-equal pairs in sorted order */
    writer_rle = ((uint64_t)neq_final * 10 > n) ? 1 : 0;
  }

  /* combiner stage: replace the sorted view with folded records */
  s->combine_applied = false;
  if (apply_combine && s->conf.combiner == 1) {
    /* partitions of the sorted records (parts_sorted, from k_eq_init) */
    static thread_local DBuf rs, rs_scan, pos2, lens2, off2, parts2, idx2;
    if (rs.alloc(8ull * n) || rs_scan.alloc(8ull * n)) return -12;
    hipLaunchKernelGGL(k_combine_mark, dim3(grid1d(n)), dim3(BLOCK), 0, 0, d_eq,
                       (uint64_t*)rs.p, n);
    uint64_t M64 = 0;
    if (scan_u64((uint64_t*)rs.p, (uint64_t*)rs_scan.p, n, &M64)) return -12;
    uint32_t M = (uint32_t)M64;
    if (pos2.alloc(4ull * M) || lens2.alloc(8ull * M) || off2.alloc(8ull * (M + 1)))
      return -12;
    hipLaunchKernelGGL(k_combine_pos, dim3(grid1d(n)), dim3(BLOCK), 0, 0, d_eq,
                       (const uint64_t*)rs_scan.p, (uint32_t*)pos2.p,
                       (uint64_t*)lens2.p, rt, d_idx, n);
    uint64_t bytes2 = 0;
    if (scan_u64((uint64_t*)lens2.p, (uint64_t*)off2.p, M, &bytes2)) return -12;
    HIP_CHECK(hipMemcpy((uint64_t*)off2.p + M, &bytes2, 8, hipMemcpyHostToDevice));
    if (s->combine_data.alloc(bytes2 ? bytes2 : 1)) return -12;
    if (s->combine_klen.alloc(4ull * M)) return -12;
    if (parts2.alloc(4ull * M) || idx2.alloc(4ull * M)) return -12;
    hipLaunchKernelGGL(k_combine_fold, dim3(grid_waves(M)), dim3(BLOCK), 0, 0, rt,
                       d_idx, (const uint32_t*)pos2.p, (const uint64_t*)off2.p,
                       (const uint32_t*)s->parts_sorted.p, (uint8_t*)s->combine_data.p,
                       (uint32_t*)s->combine_klen.p, (uint32_t*)parts2.p, M, n);
    /* swap the view: single-"spill" table over the folded records */
    std::swap(s->combine_off, off2);
    RecTable rt2 = {};
    rt2.nspills = 1;
    rt2.data0 = (const uint8_t*)s->combine_data.p;
    rt2.off0 = (const uint64_t*)s->combine_off.p;
    rt2.klen0 = (const uint32_t*)s->combine_klen.p;
    rt2.n0 = M;
    rt2.key_type = rt.key_type;
    rt = rt2;
    n = M;
    SegDesc csd;
    csd.data = rt2.data0; csd.off = rt2.off0; csd.klen = rt2.klen0;
    csd.rec_u = 0; csd.klen_u = 0;
    hsegs.assign(1, csd);
    hbase.assign(2, 0);
    hbase[1] = M;
    hipLaunchKernelGGL(k_iota, dim3(grid1d(n)), dim3(BLOCK), 0, 0,
                       (uint32_t*)idx2.p, n);
    std::swap(s->sidx, idx2);
    d_idx = (uint32_t*)s->sidx.p;
    if (s->eq.alloc(n)) return -12;
    d_eq = (uint8_t*)s->eq.p;
    HIP_CHECK(hipMemsetAsync(d_eq, 0, n));
    std::swap(s->parts_sorted, parts2);
    writer_rle = 0; /* folded keys are unique; rle never triggers */
    s->combined_parts_valid = true;
    s->combine_applied = true;
    s->combine_n = M;
    if (retain) {
      /* folded records are stored in sorted order: identity ids, composites
         rebuilt in order over the folded table (coalesced) */
      s->skey.release();
      if (retain->skey.alloc(8ull * n)) return -12;
      if (retain->sidxb.alloc(4ull * n)) return -12;
      hipLaunchKernelGGL(k_build_composite, dim3(grid1d(n)), dim3(BLOCK), 0, 0,
                         rt, (const int32_t*)s->parts_sorted.p, P, pbits,
                         ref_pb, SB, ser_mode, (uint64_t*)retain->skey.p,
                         (uint32_t*)retain->sidxb.p, n);
      if (retain->skey2.alloc(8ull * n)) return -12;
      hipLaunchKernelGGL(k_build_lkeys, dim3(grid1d(n)), dim3(BLOCK), 0, 0, rt,
                         c0, 0, ser_mode, (uint64_t*)retain->skey2.p, n);
      retain->lo_c0 = c0;
      retain->sort_sb = SB;
      retain->sort_ser_mode = ser_mode;
      retain->sorted_valid = 1;
    }
  } else {
    s->combined_parts_valid = false;
  }
  if (s->same.alloc(n)) return -12;
  static thread_local DBuf d_sprle;
  if (d_sprle.alloc(nspills_rle ? nspills_rle : 1)) return -12;
  if (nspills_rle)
    HIP_CHECK(hipMemcpy(d_sprle.p, h_spill_rle, nspills_rle, hipMemcpyHostToDevice));
  hipLaunchKernelGGL(k_writer_same, dim3(grid1d(n)), dim3(BLOCK), 0, 0, rt, d_idx, d_eq,
                     writer_rle, (const uint8_t*)d_sprle.p, (uint8_t*)s->same.p, n);

  /* uniform-emit gate: constant record serialization + RLE provably off */
  uint32_t uni_rec = 0, uni_klen = 0;
  {
    bool uni = !hsegs.empty() && !s->combine_applied;
    for (size_t sp2 = 0; sp2 < hsegs.size() && uni; sp2++) {
      if (!hsegs[sp2].rec_u || !hsegs[sp2].klen_u) uni = false;
      else if (sp2 == 0) { uni_rec = hsegs[0].rec_u; uni_klen = hsegs[0].klen_u; }
      else if (uni_rec != hsegs[sp2].rec_u || uni_klen != hsegs[sp2].klen_u)
        uni = false;
    }
    if (!uni) uni_rec = 0;
  }
  bool no_same = (writer_rle == 0) &&
      (neq_final == 0 || (nspills_rle == 1 && h_spill_rle && !h_spill_rle[0]));
  uint32_t uh_len = 0;
  uint64_t uh_word = 0;
  if (uni_rec && no_same) {
    uint8_t hb[12] = {0};
    auto hvint = [](uint8_t* b, uint32_t v) -> uint32_t {
      if (v <= 127) { b[0] = (uint8_t)v; return 1; }
      uint32_t nb2 = 0, t = v;
      while (t) { t >>= 8; nb2++; }
      b[0] = (uint8_t)(-112 - (int)nb2);
      for (uint32_t k = 0; k < nb2; k++) b[1 + k] = (uint8_t)(v >> (8 * (nb2 - 1 - k)));
      return 1 + nb2;
    };
    uh_len = hvint(hb, uni_klen);
    uh_len += hvint(hb + uh_len, uni_rec - uni_klen);
    for (int b = 0; b < 8; b++) uh_word |= (uint64_t)hb[b] << (8 * b);
    if (uh_len > 8) uni_rec = 0; /* header too long for the fast paths */
  }
  /* single-segment uniform no-RLE: src/dst/sizes are pure arithmetic — skip
     the descriptor build, the size scan AND the per-partition scan gather */
  bool arith_emit = rt.nspills == 1 && uni_rec && no_same;

  /* 5. emit sizes + partition layout (parts_sorted came from k_eq_init; the
     combiner path swapped in its own parts2) */
  static thread_local DBuf descbuf;
  uint64_t total_body = 0;
  if (!arith_emit) {
  if (descbuf.alloc(sizeof(RecDesc) * n)) return -12;
  if (s->sizes.alloc(sizeof(uint64_t) * n)) return -12;
  if (s->scan.alloc(sizeof(uint64_t) * n)) return -12;
  {
    bool direct = rt.key_type != 1;
    for (auto& sd : hsegs)
      if (!sd.rec_u) { direct = false; break; }
    if (direct) {
      /* uniform BytesWritable tables: rt_view is pure arithmetic; emit
         sizes are computed in the same pass */
      hipLaunchKernelGGL(k_build_desc, dim3(grid1d(n)), dim3(BLOCK), 0, 0, rt, d_idx,
                         (RecDesc*)descbuf.p, (const uint8_t*)s->same.p,
                         (const uint32_t*)s->parts_sorted.p,
                         (uint64_t*)s->sizes.p, n);
    } else {
      /* variable-length / Text: build in original order (coalesced off/klen/
         vint reads), then one 16B-per-record permute + fused sizes */
      static thread_local DBuf desc0;
      if (desc0.alloc(sizeof(RecDesc) * n)) return -12;
      hipLaunchKernelGGL(k_build_desc, dim3(grid1d(n)), dim3(BLOCK), 0, 0, rt,
                         (const uint32_t*)nullptr, (RecDesc*)desc0.p,
                         (const uint8_t*)nullptr, (const uint32_t*)nullptr,
                         (uint64_t*)nullptr, n);
      hipLaunchKernelGGL(k_permute_desc, dim3(grid1d(n)), dim3(BLOCK), 0, 0,
                         (const RecDesc*)desc0.p, d_idx, (RecDesc*)descbuf.p,
                         (const uint8_t*)s->same.p,
                         (const uint32_t*)s->parts_sorted.p,
                         (uint64_t*)s->sizes.p, n);
      desc0.release();
    }
  }
  if (scan_u64((uint64_t*)s->sizes.p, (uint64_t*)s->scan.p, n, &total_body)) return -12;
  s->sizes.release();
  } else {
    total_body = (uint64_t)n * (uh_len + uni_rec);
  }
  s->skey.release();  /* composites are dead once parts_sorted exists */
  /* partition record ranges: host-side from a partition histogram */
  std::vector<uint32_t> h_pcount(P, 0);
  {
    static thread_local DBuf d_pc;
    if (d_pc.alloc(sizeof(uint32_t) * P)) return -12;
    HIP_CHECK(hipMemsetAsync(d_pc.p, 0, sizeof(uint32_t) * P));
    if ((size_t)P * 4 > 64 * 1024) FAIL(-22, "num_partitions too large for r1 hist");
    hipLaunchKernelGGL(k_part_hist, dim3(grid1d(n)), dim3(BLOCK), (uint32_t)(P * 4), 0,
                       (const uint32_t*)s->parts_sorted.p, n, (uint32_t*)d_pc.p, P);
    HIP_CHECK(hipMemcpy(h_pcount.data(), d_pc.p, sizeof(uint32_t) * P,
                        hipMemcpyDeviceToHost));
  }
  std::vector<uint64_t> h_prec_start(P + 1, 0);
  for (int p = 0; p < P; p++) h_prec_start[p + 1] = h_prec_start[p] + h_pcount[p];
  s->final_rec_ranges = h_prec_start;
  s->final_rt = rt;
  s->final_n = n;
  /* body bytes per partition = scan[start of next] - scan[start] */
  /* gather per-partition scan bases + last-same flags in one kernel +
     one D2H (was 2P+1 synchronous 8-byte copies) */
  std::vector<uint64_t> h_scan_at(P + 1, 0);
  std::vector<uint8_t> h_last_same(P, 0);
  static thread_local DBuf d_pstart;
  if (d_pstart.alloc(8 * (P + 1))) return -12;
  HIP_CHECK(hipMemcpyAsync(d_pstart.p, h_prec_start.data(), 8 * (P + 1),
                           hipMemcpyHostToDevice));
  if (arith_emit) {
    for (int p = 0; p <= P; p++)
      h_scan_at[p] = h_prec_start[p] * (uint64_t)(uh_len + uni_rec);
  } else {
    static thread_local DBuf d_gathered;
    if (d_gathered.alloc(16 * (P + 1))) return -12;
    hipLaunchKernelGGL(k_gather_part_meta, dim3((P + BLOCK) / BLOCK), dim3(BLOCK), 0, 0,
                       (const uint64_t*)d_pstart.p, (const uint64_t*)s->scan.p,
                       (const uint8_t*)s->same.p, total_body, n, P,
                       (uint64_t*)d_gathered.p);
    std::vector<uint64_t> tmp(2 * (P + 1));
    HIP_CHECK(hipMemcpy(tmp.data(), d_gathered.p, 16 * (P + 1), hipMemcpyDeviceToHost));
    for (int p = 0; p <= P; p++) h_scan_at[p] = tmp[2 * p];
    for (int p = 0; p < P; p++)
      h_last_same[p] = (h_pcount[p] > 0) ? (uint8_t)tmp[2 * p + 1] : 0;
  }
  /* segment layout */
  int send_empty = s->conf.send_empty_partition_details;
  std::vector<uint64_t> h_seg_start(P, 0), h_body(P, 0), h_range_start(P, 0),
      h_range_len(P, 0), h_payload_start(P, 0), h_scanbase(P, 0);
  std::vector<uint8_t> h_present(P, 0);
  outsp->index.resize(P);
  uint64_t cursor = 0;
  for (int p = 0; p < P; p++) {
    uint64_t body = h_scan_at[p + 1] - h_scan_at[p];
    bool present = (h_pcount[p] > 0) || !send_empty;
    h_present[p] = present;
    h_seg_start[p] = cursor;
    h_body[p] = body;
    h_payload_start[p] = cursor + 4;
    h_scanbase[p] = h_scan_at[p];
    uint64_t tail = (h_last_same[p] ? 1 : 0) + 2; /* V_END? + EOF */
    int64_t rawl = 0, partl = 0;
    if (present) {
      rawl = 4 + (int64_t)body + (int64_t)tail;
      partl = rawl + 4;
      h_range_start[p] = cursor + 4;
      h_range_len[p] = body + tail;
      cursor += (uint64_t)partl;
    }
    outsp->index[p].start_offset = (int64_t)h_seg_start[p];
    outsp->index[p].raw_length = rawl;
    outsp->index[p].part_length = partl;
  }
  uint64_t stream_len = cursor;
  if (outsp->ifile.alloc(stream_len ? stream_len + 16 : 1)) return -12;
  uint8_t* d_out = (uint8_t*)outsp->ifile.p;
  outsp->ifile_len = (int64_t)stream_len;
  outsp->rle = (uint8_t)writer_rle;
  (void)hipEventRecord(ev[4]);

  /* upload layout arrays */
  static thread_local DBuf d_segstart, d_body, d_paystart, d_scanbase, d_lastsame,
      d_present, d_rstart, d_rlen, d_chunkbase, d_partcrc;
  auto up = [&](DBuf& b, const void* src, size_t sz) -> int {
    if (b.alloc(sz)) return -12;
    HIP_CHECK(hipMemcpyAsync(b.p, src, sz, hipMemcpyHostToDevice));
    return 0;
  };
  if (up(d_segstart, h_seg_start.data(), 8 * P)) return -12;
  if (up(d_body, h_body.data(), 8 * P)) return -12;
  if (up(d_paystart, h_payload_start.data(), 8 * P)) return -12;
  if (up(d_scanbase, h_scanbase.data(), 8 * P)) return -12;
  if (up(d_lastsame, h_last_same.data(), P)) return -12;
  if (up(d_present, h_present.data(), P)) return -12;
  if (up(d_rstart, h_range_start.data(), 8 * P)) return -12;
  if (up(d_rlen, h_range_len.data(), 8 * P)) return -12;

  /* 6. emit records */
  static int force_simple = -1;
  if (force_simple < 0) force_simple = getenv("TZS_EMIT_SIMPLE") ? 1 : 0;
  /* k_emit_uniform (lane-per-record word moves) measured SLOWER than the
   * half-wave path: each store instruction scatters 64 partial-line writes
   * across 64 records and write coalescing collapses (37 ms vs 10 ms at C2).
   * Kept behind TZS_EMIT_UNIFORM=1 for experiments; default off. */
  static int no_uniform_emit = -1;
  if (no_uniform_emit < 0) {
    const char* e = getenv("TZS_EMIT_UNIFORM");
    no_uniform_emit = (e && e[0] == '1') ? 0 : 1;
  }
  if (arith_emit) {
    /* force_simple falls through inside the kernel (no desc/scan exist) */
    hipLaunchKernelGGL(k_emit_uniform_arith, dim3(grid_waves(n)), dim3(BLOCK),
                       0, 0, rt.data0, d_idx,
                       (const uint32_t*)s->parts_sorted.p,
                       (const uint64_t*)d_pstart.p,
                       (const uint64_t*)d_paystart.p, d_out, n, uni_rec,
                       uh_len, uh_word);
  } else if (!no_uniform_emit && !force_simple && uni_rec && no_same &&
      uni_rec % 8 == 0 && uni_rec <= 16 * 8) {
    hipLaunchKernelGGL((k_emit_uniform<uint64_t, 16>), dim3(grid1d(n)), dim3(BLOCK),
                       0, 0, (const RecDesc*)descbuf.p, (const uint64_t*)s->scan.p,
                       (const uint32_t*)s->parts_sorted.p, (const uint64_t*)d_paystart.p,
                       (const uint64_t*)d_scanbase.p, d_out, n, uh_len, uh_word, uni_rec);
  } else if (!no_uniform_emit && !force_simple && uni_rec && no_same &&
             uni_rec % 4 == 0 && uni_rec <= 32 * 4) {
    hipLaunchKernelGGL((k_emit_uniform<uint32_t, 32>), dim3(grid1d(n)), dim3(BLOCK),
                       0, 0, (const RecDesc*)descbuf.p, (const uint64_t*)s->scan.p,
                       (const uint32_t*)s->parts_sorted.p, (const uint64_t*)d_paystart.p,
                       (const uint64_t*)d_scanbase.p, d_out, n, uh_len, uh_word, uni_rec);
  } else {
    /* k_emit_span (LDS span image + dense aligned stores) measured 29-32 ms
       vs 10.3 ms at C2 — the per-lane staging loops and LDS round trip cost
       more than the wide stores save; L2 already merges the byte path's
       consecutive stores.  Kept for experiments under TZS_EMIT_SPAN=1. */
    static int use_span = -1;
    if (use_span < 0) {
      const char* e = getenv("TZS_EMIT_SPAN");
      use_span = (e && e[0] == '1') ? 1 : 0;
    }
    static int use_v2 = -1;
    if (use_v2 < 0) {
      const char* e = getenv("TZS_EMIT_V2");
      /* measured SLOWER (emit 10.2 -> 28.7 ms at C2): the 64-record LDS
         drain serializes on shuffle+bank-conflicted byte reads; kept for
         experiments only */
      use_v2 = (e && e[0] == '1') ? 1 : 0;
    }
    if (use_v2 && !force_simple && !use_span) {
      hipLaunchKernelGGL(k_emit_records_v2, dim3(grid_waves(n)), dim3(BLOCK), 0, 0,
                         (const RecDesc*)descbuf.p,
                         (const uint8_t*)s->same.p, (const uint64_t*)s->scan.p,
                         (const uint32_t*)s->parts_sorted.p,
                         (const uint64_t*)d_paystart.p,
                         (const uint64_t*)d_scanbase.p, d_out, n);
    } else if (use_span && !force_simple) {
      hipLaunchKernelGGL(k_emit_span, dim3(grid_waves(n)), dim3(BLOCK), 0, 0,
                         (const RecDesc*)descbuf.p,
                         (const uint8_t*)s->same.p, (const uint64_t*)s->scan.p,
                         (const uint32_t*)s->parts_sorted.p,
                         (const uint64_t*)d_paystart.p,
                         (const uint64_t*)d_scanbase.p, d_out, n);
    } else {
      hipLaunchKernelGGL(k_emit_records, dim3(grid_waves(n)), dim3(BLOCK), 0, 0,
                         (const RecDesc*)descbuf.p,
                         (const uint8_t*)s->same.p, (const uint64_t*)s->scan.p,
                         (const uint32_t*)s->parts_sorted.p,
                         (const uint64_t*)d_paystart.p,
                         (const uint64_t*)d_scanbase.p, d_out, n, force_simple);
    }
  }
  (void)hipEventRecord(ev[5]);

  /* 7. CRC: chunk bases per partition */
  std::vector<uint64_t> h_chunkbase(P + 1, 0);
  for (int p = 0; p < P; p++)
    h_chunkbase[p + 1] = h_chunkbase[p] + (h_range_len[p] + CRC_CHUNK - 1) / CRC_CHUNK;
  uint64_t total_chunks = h_chunkbase[P];
  if (up(d_chunkbase, h_chunkbase.data(), 8 * (P + 1))) return -12;
  static thread_local DBuf d_chunkcrc;
  if (d_chunkcrc.alloc(4 * (total_chunks ? total_chunks : 1))) return -12;
  if (d_partcrc.alloc(4 * P)) return -12;
  /* patch headers + EOF BEFORE crc (crc covers payload..EOF) */
  hipLaunchKernelGGL(k_patch_segments, dim3((P + BLOCK - 1) / BLOCK), dim3(BLOCK), 0, 0,
                     d_out, (const uint64_t*)d_segstart.p, (const uint64_t*)d_body.p,
                     (const uint8_t*)d_lastsame.p, (const uint32_t*)d_partcrc.p,
                     (const uint8_t*)d_present.p, P);
  std::vector<uint64_t> h_scbase(P + 1, 0);
  for (int p = 0; p < P; p++) {
    uint64_t nchunks = h_chunkbase[p + 1] - h_chunkbase[p];
    h_scbase[p + 1] = h_scbase[p] + (nchunks + CRC_SC_CHUNKS - 1) / CRC_SC_CHUNKS;
  }
  uint64_t total_sc = h_scbase[P];
  static thread_local DBuf d_scbase;
  if (up(d_scbase, h_scbase.data(), 8 * (P + 1))) return -12;
  if (total_sc)
    hipLaunchKernelGGL(k_crc_chunks,
                       dim3((uint32_t)min(total_sc, (uint64_t)4096)), dim3(BLOCK), 0, 0,
                       d_out, (const uint64_t*)d_rstart.p, (const uint64_t*)d_rlen.p,
                       (const uint64_t*)d_chunkbase.p, (const uint64_t*)d_scbase.p, P,
                       (uint32_t)total_sc, (uint32_t*)d_chunkcrc.p);
  std::vector<uint64_t> h_groupbase(P + 1, 0);
  for (int p = 0; p < P; p++) {
    uint64_t nchunks = h_chunkbase[p + 1] - h_chunkbase[p];
    h_groupbase[p + 1] = h_groupbase[p] + (nchunks + CRC_GROUP_CHUNKS - 1) / CRC_GROUP_CHUNKS;
  }
  uint64_t total_groups = h_groupbase[P];
  static thread_local DBuf d_groupbase, d_groupcrc, d_grouplen;
  if (up(d_groupbase, h_groupbase.data(), 8 * (P + 1))) return -12;
  if (d_groupcrc.alloc(4 * (total_groups ? total_groups : 1))) return -12;
  if (d_grouplen.alloc(8 * (total_groups ? total_groups : 1))) return -12;
  if (total_groups)
    hipLaunchKernelGGL(k_crc_combine_groups,
                       dim3((uint32_t)min(total_groups, (uint64_t)2048)), dim3(BLOCK), 0, 0,
                       (const uint64_t*)d_rlen.p, (const uint64_t*)d_chunkbase.p,
                       (const uint64_t*)d_groupbase.p, (const uint32_t*)d_chunkcrc.p, P,
                       (uint32_t)total_groups, (uint32_t*)d_groupcrc.p,
                       (uint64_t*)d_grouplen.p);
  hipLaunchKernelGGL(k_crc_combine_final, dim3((P * WAVE + BLOCK - 1) / BLOCK),
                     dim3(BLOCK), 0, 0,
                     (const uint64_t*)d_groupbase.p, (const uint32_t*)d_groupcrc.p,
                     (const uint64_t*)d_grouplen.p, P, (uint32_t*)d_partcrc.p);
  /* re-patch to write CRC trailers (header/EOF rewrite is idempotent) */
  hipLaunchKernelGGL(k_patch_segments, dim3((P + BLOCK - 1) / BLOCK), dim3(BLOCK), 0, 0,
                     d_out, (const uint64_t*)d_segstart.p, (const uint64_t*)d_body.p,
                     (const uint8_t*)d_lastsame.p, (const uint32_t*)d_partcrc.p,
                     (const uint8_t*)d_present.p, P);
  HIP_CHECK(hipDeviceSynchronize());
  (void)hipEventRecord(ev[6]);
  (void)hipEventSynchronize(ev[6]);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, ev[2], ev[3]); T.sort_ns += (int64_t)(ms * 1e6);
  (void)hipEventElapsedTime(&ms, ev[4], ev[5]); T.emit_ns += (int64_t)(ms * 1e6);
  (void)hipEventElapsedTime(&ms, ev[5], ev[6]); T.crc_ns += (int64_t)(ms * 1e6);
  (void)hipEventElapsedTime(&ms, ev[2], ev[6]); T.total_ns += (int64_t)(ms * 1e6);
  T.dominant_kernel_ns += g_scatter_ns;
  T.sort_passes += g_scatter_launches;  /* scatter launch count (roofline) */
  T.dominant_kernel_elems += g_scatter_elems;
  for (auto& e : ev) (void)hipEventDestroy(e);
  return 0;
}

__global__ void k_max_u32(const uint32_t* a, uint32_t n, uint32_t* out) {
  uint32_t loc = 0;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x)
    loc = max(loc, a[i]);
  atomicMax(out, loc);
}
__global__ void k_part_hist(const uint32_t* parts, uint32_t n, uint32_t* counts, int P) {
  /* LDS-accumulated: 1e8 global atomics on a handful of counters serialize
     at L2 (measured 638 ms at n=1e8); block-local histogram first. */
  extern __shared__ uint32_t lh[];
  for (int j = threadIdx.x; j < P; j += blockDim.x) lh[j] = 0;
  __syncthreads();
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x)
    atomicAdd(&lh[parts[i]], 1u);
  __syncthreads();
  for (int j = threadIdx.x; j < P; j += blockDim.x)
    if (lh[j]) atomicAdd(&counts[j], lh[j]);
}
__global__ void k_check_uniform(const uint64_t* off, const uint32_t* klen, int64_t n,
                                uint64_t* mm) {
  uint64_t dmin = ~0ull, dmax = 0, kmin = ~0ull, kmax = 0;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint64_t d = off[i + 1] - off[i];
    dmin = min(dmin, d); dmax = max(dmax, d);
    uint64_t k = klen[i];
    kmin = min(kmin, k); kmax = max(kmax, k);
  }
  atomicMin((unsigned long long*)&mm[0], (unsigned long long)dmin);
  atomicMax((unsigned long long*)&mm[1], (unsigned long long)dmax);
  atomicMin((unsigned long long*)&mm[2], (unsigned long long)kmin);
  atomicMax((unsigned long long*)&mm[3], (unsigned long long)kmax);
}

__global__ void k_gather_part_meta(const uint64_t* pstart, const uint64_t* scan,
                                   const uint8_t* same, uint64_t total_body,
                                   uint32_t n, int P, uint64_t* out) {
  int p = blockIdx.x * blockDim.x + threadIdx.x;
  if (p > P) return;
  uint64_t r = pstart[p];
  out[2 * p] = (r < n) ? scan[r] : total_body;
  /* last-same of partition p (p < P): same[] at pstart[p+1]-1 */
  if (p < P) {
    uint64_t e = pstart[p + 1];
    out[2 * p + 1] = (e > 0 && e <= n && e > r) ? (uint64_t)same[e - 1] : 0;
  }
}


static int spill_impl(tzs_sorter* s, bool retain_sorted) {
  int rc = absorb_host_staging(s);
  if (rc) return rc;
  if (s->cur_n == 0) return -1; /* nothing to spill (ignoreEmptySpills) */
  SpillData* sp = new SpillData();
  /* move current buffers into the spill (columnar, will be sorted in place via
     the index; we keep records unsorted + the sorted index per spill is not
     retained — for merge we re-sort the union, so we only need the columnar
     records + emitted bytes). */
  std::swap(sp->data, s->cur_data);
  std::swap(sp->off, s->cur_off);
  std::swap(sp->klen, s->cur_klen);
  sp->n = s->cur_n;
  sp->rec_u = s->cur_rec_u;
  sp->klen_u = s->cur_klen_u;
  s->cur_first_batch = true;
  s->cur_rec_u = 0;
  s->cur_klen_u = 0;
  HostRT hrt;
  hrt.add(sp->data.p, sp->off.p, sp->klen.p, sp->rec_u, sp->klen_u,
          (uint32_t)sp->n);
  if (hrt.finish(s->conf.key_type)) return -12;
  uint8_t dummy_rle = 0;
  rc = sort_and_emit(s, hrt, (uint32_t)sp->n,
                     s->have_explicit_parts ? (const int32_t*)s->cur_part.p : nullptr,
                     &dummy_rle, 1, sp, s->conf.combiner != 0,
                     retain_sorted ? sp : nullptr);
  s->cur_part.release();
  s->cur_n = 0;
  s->cur_bytes = 0;
  if (rc) { delete sp; return rc; }
  if (s->combine_applied) {
    /* later merges must read the combined spill (reference semantics):
       swap in the folded columnar set */
    std::swap(sp->data, s->combine_data);
    std::swap(sp->off, s->combine_off);
    std::swap(sp->klen, s->combine_klen);
    sp->n = s->combine_n;
    sp->rec_u = 0;
    sp->klen_u = 0;
    s->combine_applied = false;
  }
  if (s->conf.discard_spill_streams && s->conf.final_merge_enabled) {
    sp->ifile.release();
    sp->ifile_len = 0;
  }
  s->spills.push_back(sp);
  s->ctr.spilled_records += sp->n;
  s->ctr.num_spills = (int64_t)s->spills.size();
  return (int)s->spills.size() - 1;
}

extern "C" int tzs_sorter_spill(tzs_sorter* s) { return spill_impl(s, true); }

extern "C" int tzs_sorter_num_spills(const tzs_sorter* s) { return (int)s->spills.size(); }

extern "C" int tzs_sorter_flush(tzs_sorter* s) {
  /* PipelinedSorter.flush (:665-851): final spill (forced), then final merge */
  int rc = absorb_host_staging(s);
  if (rc) return rc;
  if (s->cur_n > 0 || s->spills.empty()) {
    /* force a spill even if empty (flush forces spill(false)) */
    if (s->cur_n == 0 && s->spills.empty()) {
      /* empty spill: all partitions empty */
      SpillData* sp = new SpillData();
      sp->n = 0;
      int P = s->conf.num_partitions;
      sp->index.assign(P, tzs_index_record{0, 0, 0});
      if (!s->conf.send_empty_partition_details) {
        /* every partition gets an empty IFile (header+EOF+CRC = 10 bytes) */
        std::vector<uint8_t> host;
        static const uint8_t HDR[4] = {'T', 'I', 'F', 0};
        uint8_t eofb[2] = {0xFF, 0xFF};
        uint32_t crc = h_crc32(0, eofb, 2);
        for (int p = 0; p < P; p++) {
          sp->index[p] = tzs_index_record{(int64_t)host.size(), 6, 10};
          host.insert(host.end(), HDR, HDR + 4);
          host.push_back(0xFF); host.push_back(0xFF);
          host.push_back((uint8_t)(crc >> 24)); host.push_back((uint8_t)(crc >> 16));
          host.push_back((uint8_t)(crc >> 8)); host.push_back((uint8_t)crc);
        }
        if (sp->ifile.alloc(host.size() ? host.size() : 1)) return -12;
        HIP_CHECK(hipMemcpy(sp->ifile.p, host.data(), host.size(), hipMemcpyHostToDevice));
        sp->ifile_len = (int64_t)host.size();
      }
      s->spills.push_back(sp);
      s->ctr.num_spills = (int64_t)s->spills.size();
    } else if (s->cur_n > 0) {
      /* retention is only useful when this spill will be merged with others
         (the single-spill flush takes the rename path) */
      rc = spill_impl(s, !s->spills.empty());
      if (rc < 0) return rc;
    }
  }
  bool any_no_stream = false;
  for (auto* sp2 : s->spills)
    if (sp2->no_stream) any_no_stream = true;
  if ((!s->conf.final_merge_enabled || s->spills.size() == 1) && !any_no_stream) {
    /* rename path (flush :731-757): final output = the single spill */
    SpillData* sp = s->spills.back();
    if ((int)s->spills.size() == 1) {
      s->final_index = sp->index;
      std::swap(s->final_ifile, sp->ifile);
      s->final_len = sp->ifile_len;
      s->flushed = true;
      for (auto& ix : s->final_index) s->ctr.output_bytes_with_overhead += ix.raw_length;
      return 0;
    }
    if (!s->conf.final_merge_enabled) { s->flushed = true; return 0; }
  }
  /* Final merge (PipelinedSorter.flush final merge, :759-851; TezMerger
   * MergeQueue semantics, :466-706): every spill retained its refined sorted
   * order (skey = masked composites, sidxb = original ids), so the flush
   * merges the k sorted streams with the stable pairwise merge-path kernel
   * over 128-bit (composite, lo) keys — ceil(log2 k) passes of ~40 B/element
   * — instead of re-sorting the union
   * (~8 radix passes).  Ties go to the lower spill, so equal-composite runs
   * come out in global-id order: bit-identical to the stable union re-sort
   * the refinement + emit stages were parity-proven on.
   * Any k is supported (device-side segment descriptors; no coalesce). */
  int nsp = (int)s->spills.size();
  int orig_nsp = nsp;  /* the combiner gate follows the ORIGINAL spill count */
  /* the merge reads the columnar record sets only; the per-spill IFile
     streams are dead weight now (the reference deletes spill files after the
     final merge, PipelinedSorter.java:844-849) — free them up front so the
     1e9-record C3 shape fits in HBM */
  for (auto* sp2 : s->spills) {
    sp2->ifile.release();
    sp2->ifile_len = 0;
  }
  s->final_hrt.reset();
  HostRT& hrt = s->final_hrt;
  uint64_t total_n = 0;
  std::vector<uint8_t> spill_rle;
  std::vector<SpillData*> live;
  for (int i = 0; i < nsp; i++) {
    SpillData* sp = s->spills[i];
    if (sp->n == 0) continue;  /* flush-forced empty spill adds nothing */
    hrt.add(sp->xdata ? sp->xdata : sp->data.p,
            sp->xoff ? (const void*)sp->xoff : sp->off.p,
            sp->xklen ? (const void*)sp->xklen : sp->klen.p,
            sp->rec_u, sp->klen_u, (uint32_t)sp->n);
    total_n += (uint64_t)sp->n;
    spill_rle.push_back(sp->rle);
    live.push_back(sp);
  }
  if (total_n >> 32) FAIL(-22, "too many records for final merge");
  if (hrt.finish(s->conf.key_type)) return -12;
  int lsp = (int)live.size();
  if (lsp == 0) {
    /* all spills empty: emit the empty layout via the sort path */
    SpillData emptysp;
    rc = sort_and_emit(s, hrt, 0, nullptr, nullptr, 0, &emptysp, false);
    if (rc) return rc;
    s->final_index = emptysp.index;
    std::swap(s->final_ifile, emptysp.ifile);
    s->final_len = emptysp.ifile_len;
    s->flushed = true;
    return 0;
  }
  bool combine_at_merge = s->conf.combiner != 0 &&
                          orig_nsp >= s->conf.min_spills_for_combine;
  bool all_sorted = true;
  for (auto* sp : live)
    if (!sp->sorted_valid) all_sorted = false;
  static int force_resort = -1;
  if (force_resort < 0) force_resort = getenv("TZS_MERGE_RESORT") ? 1 : 0;
  SpillData finalsp;
  if (!all_sorted || force_resort) {
    /* fallback: stable radix re-sort of the union (round-1 path).  It
       recomputes HashPartitioner placement, so explicit partitions need
       the merge path's retained composites — fail loudly, never mis-place */
    if (s->have_explicit_parts && lsp > 1)
      FAIL(-22, "explicit-partition multi-spill merge requires the merge "
                "path (retained sorted spills); unset TZS_MERGE_RESORT");
    rc = sort_and_emit(s, hrt, (uint32_t)total_n, nullptr, spill_rle.data(),
                       lsp, &finalsp, combine_at_merge);
    if (rc) return rc;
  } else {
    auto t0 = std::chrono::steady_clock::now();
    g_scatter_ns = 0; g_scatter_launches = 0; g_scatter_elems = 0;
    /* common composite parameters across spills */
    SortParams prm = derive_sort_params(s, hrt.segs, (uint32_t)total_n);
    int common_ser = prm.ser_mode;
    for (auto* sp : live)
      if (sp->sort_ser_mode) common_ser = 1;
    int SBc = 8;
    for (auto* sp : live)
      if (sp->sort_sb < SBc) SBc = sp->sort_sb;
    uint64_t mask_c = (SBc >= 8) ? ~0ull : ~0ull << (8 * (8 - SBc));
    /* common refinement start for the lo keys */
    int c0c = common_ser ? (8 * SBc - s->pbits - prm.proxy_w) / 8
                         : (8 * SBc - s->pbits) / 8;
    if (c0c < 0) c0c = 0;
    /* spills whose retained composites used the content form must be
       rebuilt in serialized form when any spill needs it (their sorted
       order is unchanged: content order == serialized order for the
       uniform-klen spills that chose the content form — DESIGN.md §3);
       spills whose lo byte range differs from the common one rebuild the
       lo keys the same way.  (A fully-refined stream is sorted under ANY
       prefix projection of the comparator order, so rebuilding values
       never breaks per-stream sortedness.) */
    for (size_t i = 0; i < live.size(); i++) {
      SpillData* sp = live[i];
      bool ser_rebuild = common_ser && !sp->sort_ser_mode;
      bool lo_rebuild = ser_rebuild || sp->lo_c0 != c0c;
      if (!ser_rebuild && !lo_rebuild) continue;
      HostRT one;
      one.add(sp->xdata ? sp->xdata : sp->data.p,
              sp->xoff ? (const void*)sp->xoff : sp->off.p,
              sp->xklen ? (const void*)sp->xklen : sp->klen.p,
              sp->rec_u, sp->klen_u, (uint32_t)sp->n);
      if (one.finish(s->conf.key_type)) return -12;
      static thread_local DBuf rebuilt_idx, ser0;
      if (ser0.alloc(8ull * sp->n)) return -12;
      if (ser_rebuild) {
        if (rebuilt_idx.alloc(4ull * sp->n)) return -12;
        /* build ser composites in ORIGINAL order, then gather into the
           retained sorted order through sidxb */
        hipLaunchKernelGGL(k_build_composite, dim3(grid1d(sp->n)), dim3(BLOCK),
                           0, 0, one.rt, (const int32_t*)nullptr,
                           s->conf.num_partitions, s->pbits, prm.ref_pb, SBc,
                           1, (uint64_t*)ser0.p, (uint32_t*)rebuilt_idx.p,
                           (uint32_t)sp->n);
        uint64_t himask = s->pbits ? (~0ull << (64 - s->pbits)) : 0ull;
        hipLaunchKernelGGL(k_gather_merge_hi, dim3(grid1d(sp->n)), dim3(BLOCK),
                           0, 0, (const uint64_t*)ser0.p,
                           (const uint32_t*)sp->sidxb.p,
                           (uint64_t*)sp->skey.p, himask, (uint32_t)sp->n);
        sp->sort_ser_mode = 1;
        sp->sort_sb = SBc;
      }
      if (lo_rebuild) {
        hipLaunchKernelGGL(k_build_lkeys, dim3(grid1d(sp->n)), dim3(BLOCK),
                           0, 0, one.rt, c0c, 0, common_ser,
                           (uint64_t*)ser0.p, (uint32_t)sp->n);
        hipLaunchKernelGGL(k_gather_merge_hi, dim3(grid1d(sp->n)), dim3(BLOCK),
                           0, 0, (const uint64_t*)ser0.p,
                           (const uint32_t*)sp->sidxb.p,
                           (uint64_t*)sp->skey2.p, 0ull, (uint32_t)sp->n);
        sp->lo_c0 = c0c;
      }
    }
    /* merge tree: stable pairwise rounds over (hi, lo) 128-bit keys,
       final round lands in (skey, skey_lo, sidx) */
    struct MStream {
      const uint64_t* k; const uint64_t* l; const uint32_t* pidx; uint32_t n;
      uint64_t mask; uint32_t add;
      DBuf* own_k; DBuf* own_l; DBuf* own_p;
    };
    std::vector<MStream> streams;
    for (size_t i = 0; i < live.size(); i++) {
      SpillData* sp = live[i];
      MStream m;
      m.k = (const uint64_t*)sp->skey.p;
      m.l = (const uint64_t*)sp->skey2.p;
      m.pidx = (const uint32_t*)sp->sidxb.p;
      m.n = (uint32_t)sp->n;
      m.mask = mask_c;
      m.add = hrt.base[i];
      m.own_k = &sp->skey;
      m.own_l = &sp->skey2;
      m.own_p = &sp->sidxb;
      streams.push_back(m);
    }
    if (s->skey.alloc(8ull * total_n)) return -12;
    if (s->skey_lo.alloc(8ull * total_n)) return -12;
    if (s->sidx.alloc(4ull * total_n)) return -12;
    int R = 0;
    for (size_t m2 = streams.size(); m2 > 1; m2 = (m2 + 1) / 2) R++;
    if (R == 0) {
      MStream& m = streams[0];
      hipLaunchKernelGGL(k_apply_leaf2, dim3(grid1d(m.n)), dim3(BLOCK), 0, 0,
                         m.k, m.l, m.pidx, m.n, m.mask, m.add,
                         (uint64_t*)s->skey.p, (uint64_t*)s->skey_lo.p,
                         (uint32_t*)s->sidx.p);
      HIP_CHECK(hipDeviceSynchronize());
      if (m.own_k) { m.own_k->release(); m.own_l->release(); m.own_p->release(); }
    } else {
      /* two buffers suffice: each round reads ONLY the previous round's
         output (stray streams are materialized into the current round's
         buffer), so rounds alternate X <-> Y with the parity chosen so the
         final round lands in X = (s->skey, s->skey_lo, s->sidx). */
      DBuf& t1k = g_rs_tk64; DBuf& t1l = g_rs_tb64; DBuf& t1p = g_rs_ta0;
      if (R > 1 && (t1k.alloc(8ull * total_n) || t1l.alloc(8ull * total_n) ||
                    t1p.alloc(4ull * total_n)))
        return -12;
      for (int r = 1; r <= R; r++) {
        uint64_t* outk;
        uint64_t* outl;
        uint32_t* outp;
        if ((R - r) % 2 == 0) {
          outk = (uint64_t*)s->skey.p;
          outl = (uint64_t*)s->skey_lo.p;
          outp = (uint32_t*)s->sidx.p;
        } else {
          outk = (uint64_t*)t1k.p;
          outl = (uint64_t*)t1l.p;
          outp = (uint32_t*)t1p.p;
        }
        std::vector<MStream> next;
        uint64_t cursor = 0;
        static thread_local DBuf splitbuf;
        for (size_t i = 0; i + 1 < streams.size(); i += 2) {
          MStream &A = streams[i], &B = streams[i + 1];
          uint64_t on = (uint64_t)A.n + B.n;
          uint32_t nblk = (uint32_t)((on + MP2_TILE - 1) / MP2_TILE);
          if (splitbuf.alloc(4ull * (nblk + 1))) return -12;
          hipLaunchKernelGGL(k_mp2_partition, dim3(grid1d(nblk + 1)), dim3(BLOCK),
                             0, 0, A.k, A.l, A.n, A.mask, B.k, B.l, B.n, B.mask,
                             nblk, (uint32_t)MP2_TILE, (uint32_t*)splitbuf.p);
          hipLaunchKernelGGL(k_merge_path2, dim3(nblk), dim3(MP_BLOCK), 0, 0,
                             A.k, A.l, A.pidx, A.n, A.mask, A.add,
                             B.k, B.l, B.pidx, B.n, B.mask, B.add,
                             outk + cursor, outl + cursor, outp + cursor,
                             (const uint32_t*)splitbuf.p);
          MStream m;
          m.k = outk + cursor; m.l = outl + cursor; m.pidx = outp + cursor;
          m.n = (uint32_t)on;
          m.mask = ~0ull; m.add = 0;
          m.own_k = nullptr; m.own_l = nullptr; m.own_p = nullptr;
          next.push_back(m);
          cursor += on;
        }
        if (streams.size() & 1) {
          /* stray stream: materialize into this round's output so no source
             buffer is read across round boundaries */
          MStream& A = streams.back();
          hipLaunchKernelGGL(k_apply_leaf2, dim3(grid1d(A.n)), dim3(BLOCK), 0, 0,
                             A.k, A.l, A.pidx, A.n, A.mask, A.add,
                             outk + cursor, outl + cursor, outp + cursor);
          MStream m;
          m.k = outk + cursor; m.l = outl + cursor; m.pidx = outp + cursor;
          m.n = A.n;
          m.mask = ~0ull; m.add = 0;
          m.own_k = nullptr; m.own_l = nullptr; m.own_p = nullptr;
          next.push_back(m);
          cursor += A.n;
        }
        HIP_CHECK(hipDeviceSynchronize());
        for (auto& st : streams)
          if (st.own_k) { st.own_k->release(); st.own_l->release();
                          st.own_p->release(); }
        streams.swap(next);
      }
    }
    for (auto* sp : live) sp->sorted_valid = 0;
    s->times.merge_ns += std::chrono::duration_cast<std::chrono::nanoseconds>(
        std::chrono::steady_clock::now() - t0).count();
    rc = refine_and_emit(s, hrt.segs, hrt.base, hrt.rt, (uint32_t)total_n,
                         SBc, common_ser, spill_rle.data(), lsp, &finalsp,
                         combine_at_merge, nullptr,
                         (const uint64_t*)s->skey_lo.p);
    if (rc) return rc;
    s->skey_lo.release();
  }
  s->final_index = finalsp.index;
  std::swap(s->final_ifile, finalsp.ifile);
  s->final_len = finalsp.ifile_len;
  s->flushed = true;
  for (auto& ix : s->final_index) s->ctr.output_bytes_with_overhead += ix.raw_length;
  return 0;
}

extern "C" int tzs_sorter_output(tzs_sorter* s, const void** d_bytes, int64_t* nbytes,
                                 tzs_index_record* index) {
  if (!s->flushed) FAIL(-22, "flush first");
  if ((int)s->final_index.size() != s->conf.num_partitions)
    FAIL(-22, "no final output (final merge disabled: use spill_output)");
  if (d_bytes) *d_bytes = s->final_ifile.p;
  if (nbytes) *nbytes = s->final_len;
  if (index)
    memcpy(index, s->final_index.data(), sizeof(tzs_index_record) * s->final_index.size());
  return 0;
}

extern "C" int tzs_sorter_spill_output(tzs_sorter* s, int32_t spill_id,
                                       const void** d_bytes, int64_t* nbytes,
                                       tzs_index_record* index) {
  if (spill_id < 0 || spill_id >= (int)s->spills.size()) FAIL(-22, "bad spill id");
  SpillData* sp = s->spills[spill_id];
  if (d_bytes) *d_bytes = sp->ifile.p;
  if (nbytes) *nbytes = sp->ifile_len;
  if (index) memcpy(index, sp->index.data(), sizeof(tzs_index_record) * sp->index.size());
  return 0;
}

extern "C" int tzs_sorter_counters(const tzs_sorter* s, tzs_counters* out) {
  *out = s->ctr;
  return 0;
}
extern "C" int tzs_sorter_times(const tzs_sorter* s, tzs_times* out) {
  *out = s->times;
  return 0;
}

/* CRC32 of arbitrary device ranges (reuses the chunked CRC kernels) */
static int crc32_ranges(const uint8_t* d_stream,
                        const std::vector<uint64_t>& starts,
                        const std::vector<uint64_t>& lens,
                        std::vector<uint32_t>& out) {
  int P = (int)starts.size();
  out.assign(P, 0);
  if (P == 0) return 0;
  std::vector<uint64_t> chunkbase(P + 1, 0), scbase(P + 1, 0), groupbase(P + 1, 0);
  for (int p = 0; p < P; p++) {
    uint64_t nchunks = (lens[p] + CRC_CHUNK - 1) / CRC_CHUNK;
    chunkbase[p + 1] = chunkbase[p] + nchunks;
    scbase[p + 1] = scbase[p] + (nchunks + CRC_SC_CHUNKS - 1) / CRC_SC_CHUNKS;
    groupbase[p + 1] = groupbase[p] + (nchunks + CRC_GROUP_CHUNKS - 1) / CRC_GROUP_CHUNKS;
  }
  uint64_t total_chunks = chunkbase[P], total_sc = scbase[P], total_groups = groupbase[P];
  DBuf d_start, d_len, d_cb, d_sb, d_gb, d_cc, d_gc, d_gl, d_pc;
  auto up2 = [&](DBuf& b, const void* src2, size_t sz) -> int {
    if (b.alloc(sz)) return -12;
    HIP_CHECK(hipMemcpyAsync(b.p, src2, sz, hipMemcpyHostToDevice));
    return 0;
  };
  if (up2(d_start, starts.data(), 8 * P)) return -12;
  if (up2(d_len, lens.data(), 8 * P)) return -12;
  if (up2(d_cb, chunkbase.data(), 8 * (P + 1))) return -12;
  if (up2(d_sb, scbase.data(), 8 * (P + 1))) return -12;
  if (up2(d_gb, groupbase.data(), 8 * (P + 1))) return -12;
  if (d_cc.alloc(4 * (total_chunks ? total_chunks : 1))) return -12;
  if (d_gc.alloc(4 * (total_groups ? total_groups : 1))) return -12;
  if (d_gl.alloc(8 * (total_groups ? total_groups : 1))) return -12;
  if (d_pc.alloc(4 * P)) return -12;
  if (total_sc)
    hipLaunchKernelGGL(k_crc_chunks,
                       dim3((uint32_t)min(total_sc, (uint64_t)4096)), dim3(BLOCK), 0, 0,
                       d_stream, (const uint64_t*)d_start.p, (const uint64_t*)d_len.p,
                       (const uint64_t*)d_cb.p, (const uint64_t*)d_sb.p, P,
                       (uint32_t)total_sc, (uint32_t*)d_cc.p);
  if (total_groups)
    hipLaunchKernelGGL(k_crc_combine_groups,
                       dim3((uint32_t)min(total_groups, (uint64_t)2048)), dim3(BLOCK), 0, 0,
                       (const uint64_t*)d_len.p, (const uint64_t*)d_cb.p,
                       (const uint64_t*)d_gb.p, (const uint32_t*)d_cc.p, P,
                       (uint32_t)total_groups, (uint32_t*)d_gc.p, (uint64_t*)d_gl.p);
  hipLaunchKernelGGL(k_crc_combine_final, dim3((P * WAVE + BLOCK - 1) / BLOCK),
                     dim3(BLOCK), 0, 0, (const uint64_t*)d_gb.p,
                     (const uint32_t*)d_gc.p, (const uint64_t*)d_gl.p, P,
                     (uint32_t*)d_pc.p);
  HIP_CHECK(hipMemcpy(out.data(), d_pc.p, 4 * P, hipMemcpyDeviceToHost));
  return 0;
}

/* Compressed final output: the same partition segments re-framed as TIF\1
 * (zlib stream per segment, CRC32 over the COMPRESSED payload — the
 * reference's DefaultCodec framing, IFile.java:352-368).  rawLength keeps
 * the uncompressed accounting; partLength = 4 + zlib-stream + 4. */
extern "C" int tzs_sorter_output_compressed(tzs_sorter* s, const void** d_bytes,
                                            int64_t* nbytes,
                                            tzs_index_record* index) {
  if (!s->flushed) FAIL(-22, "flush first");
  if ((int)s->final_index.size() != s->conf.num_partitions)
    FAIL(-22, "no final output (final merge disabled: use spill_output)");
  if (ensure_device_constants()) return -70;
  int P = s->conf.num_partitions;
  const uint8_t* src = (const uint8_t*)s->final_ifile.p;
  /* per-present-segment uncompressed payload range = [start+4, start+raw) */
  std::vector<DefChunk> chunks;
  std::vector<int> seg_first(P + 1, 0);
  for (int p = 0; p < P; p++) {
    seg_first[p] = (int)chunks.size();
    const tzs_index_record& ix = s->final_index[p];
    if (ix.part_length <= 0) continue;
    uint64_t off = (uint64_t)ix.start_offset + 4;
    uint64_t len = (uint64_t)ix.raw_length - 4; /* body + tail, no CRC */
    uint64_t done = 0;
    while (done < len) {
      DefChunk ck;
      ck.in_off = off + done;
      ck.in_len = (uint32_t)min(len - done, (uint64_t)DEF_CHUNK);
      done += ck.in_len;
      ck.last = (done == len) ? 1 : 0;
      chunks.push_back(ck);
    }
  }
  seg_first[P] = (int)chunks.size();
  uint32_t nchunks = (uint32_t)chunks.size();
  std::vector<uint32_t> h_len(nchunks), h_adler(2 * nchunks);
  const uint32_t BATCH = 32768; /* 1.2 GB of slot scratch */
  DBuf d_chunks, d_slots, d_len, d_adl;
  if (d_chunks.alloc(sizeof(DefChunk) * (nchunks ? nchunks : 1))) return -12;
  if (nchunks)
    HIP_CHECK(hipMemcpyAsync(d_chunks.p, chunks.data(),
                             sizeof(DefChunk) * nchunks, hipMemcpyHostToDevice));
  if (d_len.alloc(4ull * (nchunks ? nchunks : 1))) return -12;
  if (d_adl.alloc(8ull * (nchunks ? nchunks : 1))) return -12;
  uint32_t nbatch = nchunks ? min(nchunks, BATCH) : 1;
  if (d_slots.alloc((uint64_t)nbatch * DEF_SLOT)) return -12;
  /* pass 1: compressed sizes + adler halves (single-batch runs keep the
     slots and skip the second deflate) */
  bool single_batch = nchunks <= BATCH;
  for (uint32_t b0 = 0; b0 < nchunks; b0 += BATCH) {
    uint32_t bc = min(BATCH, nchunks - b0);
    hipLaunchKernelGGL(k_deflate_chunks, dim3(min(bc, 4096u)), dim3(WAVE), 0, 0,
                       src, (const DefChunk*)d_chunks.p + b0, bc,
                       (uint8_t*)d_slots.p, (uint32_t*)d_len.p + b0,
                       (uint32_t*)d_adl.p + 2ull * b0);
    HIP_CHECK(hipMemcpy(h_len.data() + b0, (uint32_t*)d_len.p + b0, 4ull * bc,
                        hipMemcpyDeviceToHost));
  }
  if (nchunks)
    HIP_CHECK(hipMemcpy(h_adler.data(), d_adl.p, 8ull * nchunks,
                        hipMemcpyDeviceToHost));
  /* layout: per present segment TIF\1 + 0x78 0x9C + deflate + adler + CRC */
  std::vector<uint64_t> comp_seg(P, 0), seg_start(P, 0);
  uint64_t total = 0;
  for (int p = 0; p < P; p++) {
    uint64_t t = 0;
    for (int c = seg_first[p]; c < seg_first[p + 1]; c++) t += h_len[c];
    comp_seg[p] = t;
    seg_start[p] = total;
    if (s->final_index[p].part_length > 0)
      total += 4 + 2 + t + 4 + 4;
  }
  if (s->comp_ifile.alloc(total ? total + 16 : 1)) return -12;
  uint8_t* d_out2 = (uint8_t*)s->comp_ifile.p;
  /* pass 2: place chunks at their final offsets */
  std::vector<uint64_t> h_final_off(nchunks);
  {
    for (int p = 0; p < P; p++) {
      if (s->final_index[p].part_length <= 0) continue;
      uint64_t pos = seg_start[p] + 4 + 2;
      for (int c = seg_first[p]; c < seg_first[p + 1]; c++) {
        h_final_off[c] = pos;
        pos += h_len[c];
      }
    }
    static thread_local DBuf d_off2;
    if (d_off2.alloc(8ull * (nchunks ? nchunks : 1))) return -12;
    for (uint32_t b0 = 0; b0 < nchunks; b0 += BATCH) {
      uint32_t bc = min(BATCH, nchunks - b0);
      if (!single_batch)
        hipLaunchKernelGGL(k_deflate_chunks, dim3(min(bc, 4096u)), dim3(WAVE), 0, 0,
                           src, (const DefChunk*)d_chunks.p + b0, bc,
                           (uint8_t*)d_slots.p, (uint32_t*)d_len.p + b0,
                           (uint32_t*)d_adl.p + 2ull * b0);
      HIP_CHECK(hipMemcpyAsync(d_off2.p, h_final_off.data() + b0, 8ull * bc,
                               hipMemcpyHostToDevice));
      hipLaunchKernelGGL(k_deflate_gather, dim3(grid_waves(bc)), dim3(BLOCK), 0, 0,
                         (const uint8_t*)d_slots.p, (const uint32_t*)d_len.p + b0,
                         (const uint64_t*)d_off2.p, bc, d_out2);
      HIP_CHECK(hipDeviceSynchronize());
    }
  }
  /* headers + adler trailers */
  s->comp_index.assign(P, tzs_index_record{0, 0, 0});
  for (int p = 0; p < P; p++) {
    s->comp_index[p].start_offset = (int64_t)seg_start[p];
    if (s->final_index[p].part_length <= 0) continue;
    uint32_t a = 1, bsum = 0;
    for (int c = seg_first[p]; c < seg_first[p + 1]; c++) {
      /* zlib adler32 combine: a' = a + a2 - 1; b' = b + b2 + len2*(a-1) */
      uint32_t a2 = h_adler[2 * c], b2 = h_adler[2 * c + 1];
      uint64_t rem = chunks[c].in_len % 65521u;
      uint64_t nb2 = ((uint64_t)bsum + b2 + rem * (((uint64_t)a + 65520u) % 65521u)) % 65521u;
      a = (uint32_t)(((uint64_t)a + a2 + 65520u) % 65521u);
      bsum = (uint32_t)nb2;
    }
    uint8_t head[6] = {'T', 'I', 'F', 1, 0x78, 0x9C};
    HIP_CHECK(hipMemcpy(d_out2 + seg_start[p], head, 6, hipMemcpyHostToDevice));
    uint8_t atr[4] = {(uint8_t)(bsum >> 8), (uint8_t)bsum,
                      (uint8_t)(a >> 8), (uint8_t)a};
    HIP_CHECK(hipMemcpy(d_out2 + seg_start[p] + 6 + comp_seg[p], atr, 4,
                        hipMemcpyHostToDevice));
    s->comp_index[p].raw_length = s->final_index[p].raw_length;
    s->comp_index[p].part_length = (int64_t)(4 + 2 + comp_seg[p] + 4 + 4);
  }
  /* CRC32 over each compressed stream (zlib hdr .. adler) + trailer patch */
  {
    std::vector<uint64_t> cst, cln;
    std::vector<int> cmap;
    for (int p = 0; p < P; p++)
      if (s->final_index[p].part_length > 0) {
        cst.push_back(seg_start[p] + 4);
        cln.push_back(2 + comp_seg[p] + 4);
        cmap.push_back(p);
      }
    std::vector<uint32_t> crcs;
    int rc2 = crc32_ranges(d_out2, cst, cln, crcs);
    if (rc2) return rc2;
    for (size_t i = 0; i < cmap.size(); i++) {
      int p = cmap[i];
      uint32_t crc = crcs[i];
      uint8_t tr[4] = {(uint8_t)(crc >> 24), (uint8_t)(crc >> 16),
                       (uint8_t)(crc >> 8), (uint8_t)crc};
      HIP_CHECK(hipMemcpy(d_out2 + seg_start[p] + 4 + cln[i], tr, 4,
                          hipMemcpyHostToDevice));
    }
  }
  s->comp_len = (int64_t)total;
  if (d_bytes) *d_bytes = s->comp_ifile.p;
  if (nbytes) *nbytes = s->comp_len;
  if (index && P)
    memcpy(index, s->comp_index.data(), sizeof(tzs_index_record) * P);
  return 0;
}

extern "C" int tzs_sorter_write_files(tzs_sorter* s, const char* local_dir,
                                      const char* unique_id) {
  /* reference layout: <dir>/output/<uniqueId>/file.out + file.out.index
     (TezTaskOutputFiles.java:52-69, Constants.java:31-63) */
  if (!s->flushed) FAIL(-22, "flush first");
  std::string base = std::string(local_dir) + "/output/" + unique_id;
  mkdir((std::string(local_dir) + "/output").c_str(), 0755);
  mkdir(base.c_str(), 0755);
  std::string fo = base + "/file.out";
  std::vector<uint8_t> host(s->final_len);
  if (s->final_len)
    HIP_CHECK(hipMemcpy(host.data(), s->final_ifile.p, s->final_len, hipMemcpyDeviceToHost));
  FILE* f = fopen(fo.c_str(), "wb");
  if (!f) FAIL(-5, "open %s", fo.c_str());
  if (s->final_len) fwrite(host.data(), 1, host.size(), f);
  fclose(f);
  /* index: 24 bytes/partition BE + PureJavaCrc32 long (TezSpillRecord.java) */
  int P = s->conf.num_partitions;
  std::vector<uint8_t> ix(24 * P + 8);
  for (int p = 0; p < P; p++) {
    uint64_t v[3] = {(uint64_t)s->final_index[p].start_offset,
                     (uint64_t)s->final_index[p].raw_length,
                     (uint64_t)s->final_index[p].part_length};
    for (int j = 0; j < 3; j++)
      for (int b = 0; b < 8; b++)
        ix[24 * p + 8 * j + b] = (uint8_t)(v[j] >> (56 - 8 * b));
  }
  uint32_t crc = h_crc32(0, ix.data(), 24 * P);
  for (int b = 0; b < 8; b++) ix[24 * P + b] = (uint8_t)((uint64_t)crc >> (56 - 8 * b));
  std::string fi = fo + ".index";
  f = fopen(fi.c_str(), "wb");
  if (!f) FAIL(-5, "open %s", fi.c_str());
  fwrite(ix.data(), 1, ix.size(), f);
  fclose(f);
  return 0;
}

extern "C" void tzs_sorter_close(tzs_sorter* s) {
  if (!s) return;
  for (auto* sp : s->spills) { sp->release(); delete sp; }
  delete s;
}

/* Materialize the final sort as permuted columnar arrays (exchange wire).
 * Returns device pointers owned by the sorter and host record ranges
 * rec_ranges[P+1] / byte_ranges[P+1]. */
extern "C" int tzs_sorter_sorted_columnar(tzs_sorter* s,
                                          const void** d_data, const uint64_t** d_off,
                                          const uint32_t** d_klen,
                                          uint64_t* rec_ranges, uint64_t* byte_ranges) {
  if (!s->flushed || s->final_n == 0) {
    if (!s->flushed) FAIL(-22, "flush first");
    if (d_data) *d_data = nullptr;
    int P = s->conf.num_partitions;
    for (int p = 0; p <= P; p++) { rec_ranges[p] = 0; byte_ranges[p] = 0; }
    return 0;
  }
  uint32_t n = s->final_n;
  static thread_local DBuf lens;
  if (lens.alloc(8ull * n)) return -12;
  hipLaunchKernelGGL(k_sorted_reclens, dim3(grid1d(n)), dim3(BLOCK), 0, 0, s->final_rt,
                     (const uint32_t*)s->sidx.p, (uint64_t*)lens.p, n);
  if (s->col_off.alloc(8ull * (n + 1))) return -12;
  uint64_t total = 0;
  if (scan_u64((uint64_t*)lens.p, (uint64_t*)s->col_off.p, n, &total)) return -12;
  HIP_CHECK(hipMemcpy((uint64_t*)s->col_off.p + n, &total, 8, hipMemcpyHostToDevice));
  if (s->col_data.alloc(total ? total : 1)) return -12;
  if (s->col_klen.alloc(4ull * n)) return -12;
  hipLaunchKernelGGL(k_permute_records, dim3(grid_waves(n)), dim3(BLOCK), 0, 0, s->final_rt,
                     (const uint32_t*)s->sidx.p, (const uint64_t*)s->col_off.p,
                     (uint8_t*)s->col_data.p, (uint32_t*)s->col_klen.p, n);
  HIP_CHECK(hipDeviceSynchronize());
  int P = s->conf.num_partitions;
  for (int p = 0; p <= P; p++) {
    uint64_t r = s->final_rec_ranges[p];
    rec_ranges[p] = r;
    uint64_t b = total;
    if (r < n)
      HIP_CHECK(hipMemcpy(&b, (uint64_t*)s->col_off.p + r, 8, hipMemcpyDeviceToHost));
    byte_ranges[p] = b;
  }
  if (d_data) *d_data = s->col_data.p;
  if (d_off) *d_off = (const uint64_t*)s->col_off.p;
  if (d_klen) *d_klen = (const uint32_t*)s->col_klen.p;
  return 0;
}

/* ---- synthetic generation ---- */
extern "C" int tzs_generate(uint64_t seed, int64_t n, int32_t kind, int32_t klen,
                            int32_t vlen, const tzs_conf* conf, void** d_data,
                            uint64_t** d_off, uint32_t** d_klen, int32_t** d_part) {
  if (ensure_device_constants()) return -70;
  if (kind < 0 || kind > 3) FAIL(-22, "generator kind %d not implemented", kind);
  void* dd = nullptr;
  uint64_t* doff = nullptr;
  uint32_t* dkl = nullptr;
  if (pool_alloc_raw(8 * (n + 1), (void**)&doff)) return -12;
  if (pool_alloc_raw(4 * n, (void**)&dkl)) return -12;
  uint64_t rec = 0;
  if (kind == 0 || kind == 2 || kind == 3) {
    rec = 4 + (uint64_t)klen + 4 + (uint64_t)vlen;
    if (pool_alloc_raw(rec * n, &dd)) return -12;
    hipLaunchKernelGGL(k_generate_fixed, dim3(grid1d(n)), dim3(BLOCK), 0, 0, seed, n,
                       klen, vlen, (uint8_t*)dd, rec);
    hipLaunchKernelGGL(k_fill_fixed_offsets, dim3(grid1d(n + 1)), dim3(BLOCK), 0, 0,
                       doff, dkl, n, rec, (uint32_t)(4 + klen));
  } else { /* kind 1: Text keys (C3 shape) */
    hipLaunchKernelGGL(k_gen_text_lens, dim3(grid1d(n)), dim3(BLOCK), 0, 0, seed, n,
                       vlen, doff);
    uint64_t total = 0;
    if (scan_u64(doff, doff, (uint32_t)n, &total)) return -12;
    HIP_CHECK(hipMemcpy(doff + n, &total, 8, hipMemcpyHostToDevice));
    if (pool_alloc_raw(total ? total : 1, &dd)) return -12;
    hipLaunchKernelGGL(k_gen_text_fill, dim3(grid1d(n)), dim3(BLOCK), 0, 0, seed, n,
                       vlen, doff, (uint8_t*)dd, dkl);
  }
  if (d_part) {
    int32_t* dp = nullptr;
    if (pool_alloc_raw(4 * n, (void**)&dp)) return -12;
    if (kind == 2) {
      hipLaunchKernelGGL(k_range_partition, dim3(grid1d(n)), dim3(BLOCK), 0, 0,
                         (const uint8_t*)dd, rec, conf ? conf->num_partitions : 1, dp, n);
    } else if (kind == 3) {
      /* inverse-CDF LUT: partition p's share ∝ 1/(p+1) (Zipf s=1.0) */
      int P = conf ? conf->num_partitions : 1;
      std::vector<double> cdf(P + 1, 0.0);
      for (int p = 0; p < P; p++) cdf[p + 1] = cdf[p] + 1.0 / (p + 1);
      double tot = cdf[P];
      std::vector<int32_t> lut(65536);
      int p = 0;
      for (int v = 0; v < 65536; v++) {
        double x = (v + 0.5) / 65536.0 * tot;
        while (p + 1 < P && cdf[p + 1] < x) p++;
        lut[v] = p;
      }
      static thread_local DBuf dlut;
      if (dlut.alloc(4 * 65536)) return -12;
      HIP_CHECK(hipMemcpyAsync(dlut.p, lut.data(), 4 * 65536, hipMemcpyHostToDevice));
      hipLaunchKernelGGL(k_lut_partition, dim3(grid1d(n)), dim3(BLOCK), 0, 0,
                         (const uint8_t*)dd, rec, (const int32_t*)dlut.p, dp, n);
      HIP_CHECK(hipDeviceSynchronize());  /* lut is reused next call */
    } else {
      RecTable rt = {};
      rt.nspills = 1;
      rt.data0 = (const uint8_t*)dd;
      rt.off0 = doff;
      rt.klen0 = dkl;
      rt.n0 = (uint32_t)n;
      rt.key_type = (kind == 1) ? TZS_KEY_TEXT : (conf ? conf->key_type : TZS_KEY_BYTES);
      hipLaunchKernelGGL(k_hash_partition, dim3(grid1d(n)), dim3(BLOCK), 0, 0, rt,
                         conf ? conf->num_partitions : 1, dp, (uint32_t)n);
    }
    *d_part = dp;
  }
  HIP_CHECK(hipDeviceSynchronize());
  if (kind == 0 || kind == 2 || kind == 3) {
    /* fixed-stride kinds: record uniformity so a zero-copy adopt can skip
       its full-n uniformity scan */
    std::lock_guard<std::mutex> lk(pool_mu());
    uniform_hints()[doff] = UniformHint{(uint32_t)rec, (uint32_t)(4 + klen),
                                        rec * (uint64_t)n};
  }
  *d_data = dd;
  *d_off = doff;
  *d_klen = dkl;
  return 0;
}

extern "C" void tzs_free_device(void* p) { pool_free_raw(p); }

extern "C" void tzs_pool_stats(uint64_t out[4]) {
  /* {in-use bytes, held bytes, peak in-use bytes, drop-and-retry count} */
  std::lock_guard<std::mutex> lk(pool_mu());
  out[0] = g_pool_inuse;
  out[1] = g_pool_held;
  out[2] = g_pool_peak;
  out[3] = g_pool_drops;
}

/* ---- reduce-side pre-sorted segment ingestion --------------------------
 * The MI355X MergeManager path (MergeManager.java:423-519,1162-1328): a
 * fetched/exchanged columnar segment is ALREADY sorted by (partition, key)
 * — the map side sorted it — so the reduce merge must not re-sort it.  Each
 * call wraps the caller's device buffers (caller keeps them alive until
 * flush/close) as one sorted spill: composites are built in order (one
 * coalesced read), validated nondecreasing, and flush() k-way-merges the
 * segments with the merge-path tree.  d_part carries the records' original
 * partition ids (explicit partitioners ride through); null = recompute
 * HashPartitioner placement. */
extern "C" int tzs_sorter_add_sorted_segment(tzs_sorter* s, const void* d_data,
                                             const uint64_t* d_off,
                                             const uint32_t* d_klen,
                                             const int32_t* d_part, int64_t n) {
  if (n < 0) FAIL(-22, "bad n");
  if (n == 0) return 0;
  if (n > 4000000000ll) FAIL(-22, "segment too large (u32 record ids)");
  if (s->cur_n != 0 || !s->host_klen.empty())
    FAIL(-22, "add_sorted_segment cannot mix with buffered writes");
  uint64_t first = 0, nbytes = 0;
  HIP_CHECK(hipMemcpy(&first, d_off, 8, hipMemcpyDeviceToHost));
  HIP_CHECK(hipMemcpy(&nbytes, d_off + n, 8, hipMemcpyDeviceToHost));
  if (first != 0) FAIL(-22, "d_off must start at 0");
  nbytes -= first;
  SpillData* sp = new SpillData();
  sp->n = n;
  sp->xdata = d_data;
  sp->xoff = d_off;
  sp->xklen = d_klen;
  sp->no_stream = 1;
  sp->rle = 0; /* raw columnar records carry no source RLE markers */
  /* uniformity (enables the arithmetic rt_view fast path) */
  {
    static thread_local DBuf mm;
    if (mm.alloc(32)) { delete sp; return -12; }
    uint64_t init[4] = {~0ull, 0, ~0ull, 0};
    HIP_CHECK(hipMemcpyAsync(mm.p, init, 32, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(k_check_uniform, dim3(grid1d(n)), dim3(BLOCK), 0, 0,
                       d_off, d_klen, n, (uint64_t*)mm.p);
    uint64_t res[4];
    HIP_CHECK(hipMemcpy(res, mm.p, 32, hipMemcpyDeviceToHost));
    bool uni = (res[0] == res[1]) && (res[2] == res[3]) && res[0] <= 0xFFFFFFFFull;
    if (getenv("TZS_NO_UNIFORM")) uni = false;
    sp->rec_u = uni ? (uint32_t)res[0] : 0;
    sp->klen_u = uni ? (uint32_t)res[2] : 0;
  }
  /* composites in segment order (sorted order == original order here) */
  HostRT one;
  one.add(d_data, d_off, d_klen, sp->rec_u, sp->klen_u, (uint32_t)n);
  if (one.finish(s->conf.key_type)) { delete sp; return -12; }
  std::vector<SegDesc> hseg1(1);
  hseg1[0] = one.segs[0];
  SortParams prm = derive_sort_params(s, hseg1, (uint32_t)n);
  if (sp->skey.alloc(8ull * n)) { delete sp; return -12; }
  if (sp->sidxb.alloc(4ull * n)) { delete sp; return -12; }
  hipLaunchKernelGGL(k_build_composite, dim3(grid1d(n)), dim3(BLOCK), 0, 0,
                     one.rt, d_part, s->conf.num_partitions, s->pbits,
                     prm.ref_pb, prm.SB, prm.ser_mode,
                     (uint64_t*)sp->skey.p, (uint32_t*)sp->sidxb.p, (uint32_t)n);
  {
    static thread_local DBuf bad;
    if (bad.alloc(4)) { delete sp; return -12; }
    HIP_CHECK(hipMemsetAsync(bad.p, 0, 4));
    hipLaunchKernelGGL(k_check_sorted, dim3(grid1d(n)), dim3(BLOCK), 0, 0,
                       (const uint64_t*)sp->skey.p, (uint32_t)n,
                       (uint32_t*)bad.p);
    uint32_t h_bad = 0;
    HIP_CHECK(hipMemcpy(&h_bad, bad.p, 4, hipMemcpyDeviceToHost));
    if (h_bad) {
      delete sp;
      FAIL(-22, "add_sorted_segment: segment is not sorted (%u inversions)",
           h_bad);
    }
  }
  {
    int c0 = prm.ser_mode ? (8 * prm.SB - s->pbits - prm.proxy_w) / 8
                          : (8 * prm.SB - s->pbits) / 8;
    if (c0 < 0) c0 = 0;
    if (sp->skey2.alloc(8ull * n)) { delete sp; return -12; }
    hipLaunchKernelGGL(k_build_lkeys, dim3(grid1d(n)), dim3(BLOCK), 0, 0,
                       one.rt, c0, 0, prm.ser_mode, (uint64_t*)sp->skey2.p,
                       (uint32_t)n);
    sp->lo_c0 = c0;
  }
  sp->sort_sb = prm.SB;
  sp->sort_ser_mode = prm.ser_mode;
  sp->sorted_valid = 1;
  if (d_part) s->have_explicit_parts = true;
  s->spills.push_back(sp);
  s->ctr.output_records += n;
  s->ctr.output_bytes += (int64_t)nbytes;
  s->ctr.num_spills = (int64_t)s->spills.size();
  return (int)s->spills.size() - 1;
}

/* ---- reduce-side merge over columnar segments ---- */
extern "C" int tzs_merge_segments(const tzs_conf* conf, const tzs_segment* segs,
                                  int32_t nsegs, void** d_out, int64_t* out_bytes,
                                  tzs_index_record* rec) {
  /* Reduce-side k-way merge of sorted columnar segments of ONE partition
   * (TezMerger.MergeQueue restated as a stable re-sort — DESIGN.md §4).
   * Each segment is absorbed (appended + offset-rebased) and flush() runs
   * the union sort + IFile emit. */
  if (!conf || nsegs < 0) FAIL(-22, "bad args");
  tzs_conf c1 = *conf;
  c1.num_partitions = 1;
  tzs_sorter* s = nullptr;
  int rc2 = tzs_sorter_create(&c1, &s);
  if (rc2) return rc2;
  for (int i = 0; i < nsegs; i++) {
    rc2 = tzs_sorter_write_batch_device(s, segs[i].d_data, segs[i].d_off,
                                        segs[i].d_klen, nullptr, segs[i].n);
    if (rc2) { tzs_sorter_close(s); return rc2; }
  }
  rc2 = tzs_sorter_flush(s);
  if (rc2) { tzs_sorter_close(s); return rc2; }
  const void* db = nullptr;
  int64_t nb = 0;
  tzs_index_record ix;
  rc2 = tzs_sorter_output(s, &db, &nb, &ix);
  if (rc2) { tzs_sorter_close(s); return rc2; }
  void* own = nullptr;
  if (nb) {
    if (hipMalloc(&own, (size_t)nb) != hipSuccess) { tzs_sorter_close(s); FAIL(-12, "oom"); }
    if (hipMemcpy(own, db, (size_t)nb, hipMemcpyDeviceToDevice) != hipSuccess) {
      (void)hipFree(own); tzs_sorter_close(s); FAIL(-70, "copy");
    }
  }
  *d_out = own;
  *out_bytes = nb;
  if (rec) *rec = ix;
  tzs_sorter_close(s);
  return 0;
}

/* ---- misc helpers for the Python layer ---- */
extern "C" int tzs_memcpy_d2h(void* host, const void* dev, uint64_t n) {
  HIP_CHECK(hipMemcpy(host, dev, n, hipMemcpyDeviceToHost));
  return 0;
}
extern "C" int tzs_memcpy_d2d(void* dst, const void* src, uint64_t n) {
  HIP_CHECK(hipMemcpy(dst, src, n, hipMemcpyDeviceToDevice));
  return 0;
}
extern "C" int tzs_memcpy_h2d(void* dev, const void* host, uint64_t n) {
  HIP_CHECK(hipMemcpy(dev, host, n, hipMemcpyHostToDevice));
  return 0;
}
extern "C" int tzs_malloc_device(uint64_t n, void** out) {
  if (pool_alloc_raw(n, out)) return -12;
  return 0;
}
extern "C" int tzs_device_available(void) {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n > 0;
}
