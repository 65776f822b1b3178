/*
 * gpuq.hip — MI355X-native (gfx950, CDNA4) kernels behind the gpuq C-ABI.
 *
 * Everything here is HBM-bandwidth-bound integer/hash work (no MFMA — there
 * is no dense contraction on this path): design centers on
 * coalesced 64-lane access, LDS digit histograms, wave-wide ballot
 * multi-split ranking, and device-scope atomics.
 *
 * Reference semantics restated (cites into /root/reference):
 *  - LSB radix sort, 8-bit digits, skip-uniform-bytes:
 *    core/.../unsafe/sort/RadixSort.java:43-139 — here as a stable
 *    three-kernel pass (per-block histogram, global exclusive scan over
 *    [bin][block], ranked scatter through LDS staging).
 *  - sort-key encodings: PrefixComparators.java:66-83 (double bijection),
 *    SignedPrefixComparator (two's-complement order == unsigned order with
 *    the sign bit flipped). Descending = stable ascending radix on the
 *    bitwise complement (order-equivalent to RadixSort's desc bucket walk).
 *  - Murmur3_x86_32: common/unsafe/.../hash/Murmur3_x86_32.java:45-147.
 *  - partition id: Pmod(Murmur3Hash(key,42), n), partitioning.scala:328-330.
 *  - hash aggregate / hash join: observable semantics of
 *    HashAggregateExec + BytesToBytesMap and ShuffledHashJoinExec +
 *    LongHashedRelation (open addressing; duplicates chained) — GPU-native
 *    linear probing with device-scope atomics, not a translation.
 */
#include <hip/hip_runtime.h>
#include <stdint.h>
#include <stdio.h>
#include <string.h>

#include "../../include/gpuq.h"

/* ================= error machinery ================= */

static __thread char g_err[512];

extern "C" const char* gpuq_last_error(void) { return g_err; }

#define FAIL(code, ...) do { \
    snprintf(g_err, sizeof(g_err), __VA_ARGS__); return (code); } while (0)

#define HIP_TRY(expr) do { hipError_t _e = (expr); if (_e != hipSuccess) { \
    snprintf(g_err, sizeof(g_err), "%s failed: %s (%s:%d)", #expr, \
             hipGetErrorString(_e), __FILE__, __LINE__); \
    return GPUQ_ERR_HIP; } } while (0)

extern "C" int gpuq_device_count(void) {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n;
}

/* ---- optional per-kernel HIP-event profiling (bench roofline) ----
 * Events are recorded on the launching stream around individual kernel
 * launches; gpuq_kernel_stats() drains (synchronizes) and accumulates. */
#include <vector>
#include <map>
#include <string>
#include <mutex>

static std::mutex g_prof_mu;
static bool g_prof_on = false;
struct prof_rec { const char* tag; hipEvent_t a, b; };
static std::vector<prof_rec> g_prof_pending;
static std::map<std::string, std::pair<double, long long>> g_prof_acc;

extern "C" void gpuq_profiling(int enable) { g_prof_on = enable != 0; }

static hipEvent_t prof_begin(hipStream_t s) {
  if (!g_prof_on) return nullptr;
  hipEvent_t e; (void)hipEventCreate(&e); (void)hipEventRecord(e, s);
  return e;
}
static void prof_end(const char* tag, hipStream_t s, hipEvent_t a) {
  if (!a) return;
  hipEvent_t b; (void)hipEventCreate(&b); (void)hipEventRecord(b, s);
  std::lock_guard<std::mutex> g(g_prof_mu);
  g_prof_pending.push_back({tag, a, b});
}

extern "C" void gpuq_kernel_stats_reset(void) {
  std::lock_guard<std::mutex> g(g_prof_mu);
  for (auto& r : g_prof_pending) { (void)hipEventDestroy(r.a); (void)hipEventDestroy(r.b); }
  g_prof_pending.clear();
  g_prof_acc.clear();
}

extern "C" int gpuq_kernel_stats(const char* name, double* total_ms, long long* count) {
  std::lock_guard<std::mutex> g(g_prof_mu);
  for (auto& r : g_prof_pending) {
    (void)hipEventSynchronize(r.b);
    float ms = 0;
    (void)hipEventElapsedTime(&ms, r.a, r.b);
    auto& acc = g_prof_acc[r.tag];
    acc.first += ms; acc.second += 1;
    (void)hipEventDestroy(r.a); (void)hipEventDestroy(r.b);
  }
  g_prof_pending.clear();
  auto it = g_prof_acc.find(name);
  if (it == g_prof_acc.end()) { *total_ms = 0; *count = 0; return GPUQ_OK; }
  *total_ms = it->second.first;
  *count = it->second.second;
  return GPUQ_OK;
}

/* ================= device inlines ================= */

#define WAVE 64
#define DEV static __device__ __forceinline__

DEV uint32_t rotl32(uint32_t x, int r) { return (x << r) | (x >> (32 - r)); }

/* Murmur3_x86_32.java:125-147 */
DEV uint32_t mm3_mixK1(uint32_t k1) {
  k1 *= 0xcc9e2d51u; k1 = rotl32(k1, 15); k1 *= 0x1b873593u; return k1;
}
DEV uint32_t mm3_mixH1(uint32_t h1, uint32_t k1) {
  h1 ^= k1; h1 = rotl32(h1, 13); return h1 * 5u + 0xe6546b64u;
}
DEV uint32_t mm3_fmix(uint32_t h1, uint32_t len) {
  h1 ^= len; h1 ^= h1 >> 16; h1 *= 0x85ebca6bu; h1 ^= h1 >> 13;
  h1 *= 0xc2b2ae35u; h1 ^= h1 >> 16; return h1;
}
/* Murmur3_x86_32.java:109-122 hashLong */
DEV int32_t mm3_hash_long(int64_t input, int32_t seed) {
  uint32_t lo = (uint32_t)(uint64_t)input;
  uint32_t hi = (uint32_t)((uint64_t)input >> 32);
  uint32_t h1 = mm3_mixH1((uint32_t)seed, mm3_mixK1(lo));
  h1 = mm3_mixH1(h1, mm3_mixK1(hi));
  return (int32_t)mm3_fmix(h1, 8);
}
/* Pmod (catalyst arithmetic.scala Pmod.pmod for int) */
DEV int32_t spark_pmod(int32_t a, int32_t n) {
  int32_t r = a % n; return r < 0 ? r + n : r;
}

/* splitmix64 — bit-identical to oracle/oracle.c gen_u64 */
DEV uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ULL;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
  return x ^ (x >> 31);
}
DEV uint64_t gen_u64_dev(uint64_t seed, uint64_t i) {
  return splitmix64(seed * 0x9E3779B97F4A7C15ULL + i);
}

#define SIGNBIT 0x8000000000000000ULL

/* PrefixComparators.java:72-83 DoublePrefixComparator.computePrefix */
DEV uint64_t encode_f64(double v) {
  if (v == -0.0) v = 0.0;
  uint64_t bits;
  if (v != v) bits = 0x7ff8000000000000ULL;  /* Java canonical NaN */
  else bits = __double_as_longlong(v);
  uint64_t mask = (uint64_t)(-(int64_t)(bits >> 63)) | SIGNBIT;
  return bits ^ mask;
}
/* SignedPrefixComparator order == unsigned order with sign flipped */
DEV uint64_t encode_i64(int64_t v) { return (uint64_t)v ^ SIGNBIT; }

/* inverses of the monotone encodings (also used by MIN/MAX accumulators,
 * which hold encoded u64 so atomicMin/atomicMax give the right order) */
DEV int64_t decode_i64(uint64_t e) { return (int64_t)(e ^ SIGNBIT); }
DEV double decode_f64(uint64_t e) {
  uint64_t mask = ((e >> 63) ? 0 : ~0ULL) | SIGNBIT;
  return __longlong_as_double((long long)(e ^ mask));
}

DEV bool bit_valid(const uint8_t* validity, int64_t i) {
  return !validity || ((validity[i >> 3] >> (i & 7)) & 1);
}

/* ================= synthetic data generation ================= */

__global__ void k_gen_i64_range(uint64_t seed, uint64_t start, int64_t n,
                                uint64_t range, int64_t* out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    uint64_t v = gen_u64_dev(seed, start + (uint64_t)i);
    out[i] = (int64_t)(range ? v % range : v);
  }
}

__global__ void k_gen_f64_unit(uint64_t seed, uint64_t start, int64_t n, double* out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    uint64_t v = gen_u64_dev(seed, start + (uint64_t)i);
    out[i] = (double)(v >> 11) * (1.0 / 9007199254740992.0);
  }
}

/* grid shape for random-access hash kernels (A/B via GPUQ_HASH_GRID:
 * 0 = one element per thread, N = cap at N blocks with grid-stride) */
static dim3 hash_grid(int64_t n, int block = 256) {
  static int cap = -2;
  if (cap == -2) {
    const char* e = getenv("GPUQ_HASH_GRID");
    cap = e ? atoi(e) : 2048;
  }
  int64_t b = (n + block - 1) / block;
  if (cap > 0 && b > cap) b = cap;
  if (b > 0x7FFFFFFF) b = 0x7FFFFFFF;
  if (b < 1) b = 1;
  return dim3((uint32_t)b);
}

static dim3 grid1d(int64_t n, int block = 256) {
  int64_t b = (n + block - 1) / block;
  if (b > 2048) b = 2048;  /* grid-stride beyond (G11: cap + stride) */
  if (b < 1) b = 1;
  return dim3((uint32_t)b);
}

extern "C" int gpuq_gen_i64_range(void* stream, uint64_t seed, uint64_t start,
                                  int64_t n, uint64_t range, int64_t* out) {
  k_gen_i64_range<<<grid1d(n), 256, 0, (hipStream_t)stream>>>(seed, start, n, range, out);
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

extern "C" int gpuq_gen_f64_unit(void* stream, uint64_t seed, uint64_t start,
                                 int64_t n, double* out) {
  k_gen_f64_unit<<<grid1d(n), 256, 0, (hipStream_t)stream>>>(seed, start, n, out);
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

/* ================= gather ================= */

template <typename T>
__global__ void k_gather(int64_t n, const T* in, const uint32_t* perm, T* out) {
  /* 4-way unrolled so four independent random loads are in flight per lane
   * (random gathers are latency-bound; same lesson as the scatter preload) */
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i + 3 * stride < n; i += 4 * stride) {
    uint32_t p0 = perm[i], p1 = perm[i + stride], p2 = perm[i + 2 * stride],
             p3 = perm[i + 3 * stride];
    T v0 = in[p0], v1 = in[p1], v2 = in[p2], v3 = in[p3];
    out[i] = v0; out[i + stride] = v1;
    out[i + 2 * stride] = v2; out[i + 3 * stride] = v3;
  }
  for (; i < n; i += stride) out[i] = in[perm[i]];
}

/* two columns through one permutation (both payload gathers of a sort step
 * in one kernel: one perm read, two independent random loads in flight) */
__global__ void k_gather2(int64_t n, const uint64_t* a, const uint64_t* b,
                          const uint32_t* perm, uint64_t* oa, uint64_t* ob) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i + stride < n; i += 2 * stride) {
    uint32_t p0 = perm[i], p1 = perm[i + stride];
    uint64_t a0 = a[p0], b0 = b[p0], a1 = a[p1], b1 = b[p1];
    oa[i] = a0; ob[i] = b0;
    oa[i + stride] = a1; ob[i + stride] = b1;
  }
  for (; i < n; i += stride) {
    uint32_t p = perm[i];
    oa[i] = a[p]; ob[i] = b[p];
  }
}

extern "C" int gpuq_gather2_i64(void* stream, int64_t n, const void* a,
                                const void* b, const uint32_t* perm,
                                void* oa, void* ob) {
  { hipEvent_t _pe = prof_begin((hipStream_t)stream);
  k_gather2<<<grid1d(n), 256, 0, (hipStream_t)stream>>>(
      n, (const uint64_t*)a, (const uint64_t*)b, perm,
      (uint64_t*)oa, (uint64_t*)ob);
  prof_end("gather2", (hipStream_t)stream, _pe); }
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

/* ---- interleaved two-column gather ----
 * The plain two-column gather issues 2 random 8 B loads per row; each
 * touches a 64 B line, so the fetch amplification is ~5x and the kernel
 * saturates on request rate (round-1 PMC: 39 B fetched per 8 B row).
 * Interleaving the source columns into 16 B records first (one streaming
 * pass) halves the random request count and doubles bytes-per-line used:
 * one b128 load serves both columns. */

__global__ void k_interleave2(int64_t n, const uint64_t* a, const uint64_t* b,
                              ulonglong2* out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += gs) {
    ulonglong2 v;
    v.x = a[i];
    v.y = b[i];
    out[i] = v;
  }
}

__global__ void k_gather2_pairs(int64_t n, const ulonglong2* pairs,
                                const uint32_t* perm, uint64_t* oa,
                                uint64_t* ob) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += gs) {
    ulonglong2 v = pairs[perm[i]];
    oa[i] = v.x;
    ob[i] = v.y;
  }
}

/* 4-way batched variant: each thread keeps 4 independent b128 loads in
 * flight (A/B vs the scalar grid-stride loop; selected by env). */
__global__ void k_gather2_pairs4(int64_t n, const ulonglong2* pairs,
                                 const uint32_t* perm, uint64_t* oa,
                                 uint64_t* ob) {
  int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t base = t * 4; base < n; base += stride) {
    uint32_t p[4];
    ulonglong2 v[4];
    int top = (int)((n - base) < 4 ? (n - base) : 4);
    #pragma unroll
    for (int q = 0; q < 4; q++)
      if (q < top) p[q] = perm[base + q];
    #pragma unroll
    for (int q = 0; q < 4; q++)
      if (q < top) v[q] = pairs[p[q]];
    #pragma unroll
    for (int q = 0; q < 4; q++)
      if (q < top) {
        oa[base + q] = v[q].x;
        ob[base + q] = v[q].y;
      }
  }
}

extern "C" int gpuq_interleave2_i64(void* stream, int64_t n, const void* a,
                                    const void* b, void* pairs) {
  hipStream_t s = (hipStream_t)stream;
  if (n == 0) return GPUQ_OK;
  { hipEvent_t _pe = prof_begin(s);
  k_interleave2<<<grid1d(n), 256, 0, s>>>(n, (const uint64_t*)a,
                                          (const uint64_t*)b,
                                          (ulonglong2*)pairs);
  prof_end("interleave2", s, _pe); }
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

extern "C" int gpuq_gather2_pairs(void* stream, int64_t n, const void* pairs,
                                  const uint32_t* perm, void* oa, void* ob) {
  hipStream_t s = (hipStream_t)stream;
  if (n == 0) return GPUQ_OK;
  static int mlp0 = -1;
  if (mlp0 < 0) mlp0 = getenv("GPUQ_GATHER_MLP") ? atoi(getenv("GPUQ_GATHER_MLP")) : 0;
  { hipEvent_t _pe = prof_begin(s);
  if (mlp0)
    k_gather2_pairs4<<<grid1d((n + 3) / 4), 256, 0, s>>>(
        n, (const ulonglong2*)pairs, perm, (uint64_t*)oa, (uint64_t*)ob);
  else
    k_gather2_pairs<<<grid1d(n), 256, 0, s>>>(n, (const ulonglong2*)pairs,
                                              perm, (uint64_t*)oa,
                                              (uint64_t*)ob);
  prof_end("gather2_pairs", s, _pe); }
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

/* scratch: n * 16 bytes, device. */
extern "C" int gpuq_gather2_i64_fast(void* stream, int64_t n, const void* a,
                                     const void* b, const uint32_t* perm,
                                     void* oa, void* ob, void* scratch) {
  hipStream_t s = (hipStream_t)stream;
  if (n == 0) return GPUQ_OK;
  { hipEvent_t _pe = prof_begin(s);
  k_interleave2<<<grid1d(n), 256, 0, s>>>(n, (const uint64_t*)a,
                                          (const uint64_t*)b,
                                          (ulonglong2*)scratch);
  prof_end("interleave2", s, _pe); }
  HIP_TRY(hipGetLastError());
  static int mlp = -1;
  if (mlp < 0) mlp = getenv("GPUQ_GATHER_MLP") ? atoi(getenv("GPUQ_GATHER_MLP")) : 0;
  { hipEvent_t _pe = prof_begin(s);
  if (mlp)
    k_gather2_pairs4<<<grid1d((n + 3) / 4), 256, 0, s>>>(
        n, (const ulonglong2*)scratch, perm, (uint64_t*)oa, (uint64_t*)ob);
  else
    k_gather2_pairs<<<grid1d(n), 256, 0, s>>>(n, (const ulonglong2*)scratch,
                                              perm, (uint64_t*)oa,
                                              (uint64_t*)ob);
  prof_end("gather2_pairs", s, _pe); }
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

extern "C" int gpuq_gather(void* stream, int64_t n, gpuq_col col,
                           const uint32_t* perm, void* out) {
  hipStream_t s = (hipStream_t)stream;
  if (col.dtype == GPUQ_INT64 || col.dtype == GPUQ_FLOAT64) {
    k_gather<uint64_t><<<grid1d(n, 256), 256, 0, s>>>(n, (const uint64_t*)col.data, perm, (uint64_t*)out);
  } else if (col.dtype == GPUQ_INT32) {
    k_gather<uint32_t><<<grid1d(n, 256), 256, 0, s>>>(n, (const uint32_t*)col.data, perm, (uint32_t*)out);
  } else {
    FAIL(GPUQ_ERR_INVALID, "gather: unsupported dtype %d", col.dtype);
  }
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

/* ================= radix sort / partition machinery ================= */
/*
 * Stable LSB radix over 8-bit digits on (u64 key, u32 rowid) pairs.
 * Per pass: per-block 256-bin histogram -> global exclusive scan over the
 * [bin][block] matrix (bin-major => stable (bin, block, in-block) order) ->
 * ranked scatter. In-block stable ranks come from a wave-level ballot
 * multi-split (8 ballots reconstruct the same-digit lane mask), wave-private
 * LDS counters, then a cross-wave/cross-bin LDS scan; the tile is staged
 * reordered through LDS so global writes go out as contiguous digit runs
 * (coalesced except run boundaries).
 */

#define SORT_BLOCK 256
#define SORT_WAVES (SORT_BLOCK / WAVE)
#define SORT_ITEMS 16
#define SORT_TILE (SORT_BLOCK * SORT_ITEMS) /* 4096 */

/* encode kernel: keys -> radix-encoded u64 + identity rowids + bitwise
 * AND/OR reduction for the skip-uniform-byte decision (RadixSort.java:113-124) */
/* encode + identity rowids + skip-byte AND/OR reduction + ALL EIGHT global
 * byte histograms in one read (one pass replaces the reference's per-byte
 * count loops, RadixSort.java:126-135; per-byte histograms are invariant
 * under the passes' permutations, so one upfront count serves every pass). */
template <int DTYPE, bool DESC>
__global__ void k_encode(int64_t n, const void* keys, const uint32_t* rowmap,
                         uint64_t* ek, uint32_t* idx,
                         unsigned long long* bits_and, unsigned long long* bits_or,
                         uint32_t* ghist /* [8][256] */) {
  __shared__ uint32_t h[8][256];
  for (int b = threadIdx.x; b < 8 * 256; b += blockDim.x) ((uint32_t*)h)[b] = 0;
  __syncthreads();
  uint64_t acc_or = 0, acc_and = ~0ULL;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    uint64_t e;
    uint32_t row = rowmap ? rowmap[i] : (uint32_t)i;
    if (DTYPE == GPUQ_FLOAT64) e = encode_f64(((const double*)keys)[row]);
    else e = encode_i64(((const int64_t*)keys)[row]);
    if (DESC) e = ~e;
    ek[i] = e;
    idx[i] = row;
    acc_or |= e; acc_and &= e;
    #pragma unroll
    for (int b = 0; b < 8; b++) atomicAdd(&h[b][(e >> (b * 8)) & 0xff], 1u);
  }
  /* wave reduce then one atomic per wave (G12) */
  for (int off = 32; off > 0; off >>= 1) {
    acc_or |= __shfl_down((unsigned long long)acc_or, off);
    acc_and &= __shfl_down((unsigned long long)acc_and, off);
  }
  if ((threadIdx.x & (WAVE - 1)) == 0) {
    atomicOr(bits_or, (unsigned long long)acc_or);
    atomicAnd(bits_and, (unsigned long long)acc_and);
  }
  __syncthreads();
  for (int b = threadIdx.x; b < 8 * 256; b += blockDim.x) {
    uint32_t v = ((uint32_t*)h)[b];
    if (v) atomicAdd(&ghist[b], v);
  }
}

/* ---- ranked scatter pass ----
 * BIN_MODE 0: digit = (key >> shift) & 0xff  (radix sort pass)
 * BIN_MODE 2: digit = top 8 bits of Murmur3(key,42) (hash-order
 *             bucketing for the locality-ordered hash join)             */
struct scatter_geom { int block, items; };
static scatter_geom get_sort_geom(void);

template <int BIN_MODE>
DEV int compute_bin(uint64_t key, int shift, int nparts) {
  (void)nparts;
  if (BIN_MODE == 0) return (int)((key >> shift) & 0xff);
  if (BIN_MODE == 2)
    /* top 8 bits of Murmur3(key,42) — hash-order bucketing so a bucketed
     * stream sweeps a hash-ordered table monotonically (join) */
    return (int)(((uint32_t)mm3_hash_long((int64_t)key, 42)) >> 24);
  /* BIN_MODE 3: byte `shift/8` of the TOP-16 murmur bits (two stable
   * passes order rows by a 16-bit hash bucket — partitioned aggregation) */
  return (int)((((uint32_t)mm3_hash_long((int64_t)key, 42)) >> (16 + shift)) & 0xff);
}

/* validity split for sort-with-nulls: pair key = 0 for the group that
 * sorts first (nulls when nulls_first), 1 for the other; also counts the
 * null rows (SortExec null ordering, SortOrder.scala:35-45) */
__global__ void k_validity_pids(int64_t n, const uint8_t* validity, int nulls_first,
                                uint64_t* pid_as_key, uint32_t* idx,
                                unsigned long long* null_count) {
  __shared__ unsigned int h;
  if (threadIdx.x == 0) h = 0;
  __syncthreads();
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  unsigned int mine = 0;
  for (; i < n; i += stride) {
    bool valid = (validity[i >> 3] >> (i & 7)) & 1;
    pid_as_key[i] = (uint64_t)(valid == (bool)nulls_first);
    idx[i] = (uint32_t)i;
    if (!valid) mine++;
  }
  if (mine) atomicAdd(&h, mine);
  __syncthreads();
  if (threadIdx.x == 0 && h) atomicAdd(null_count, (unsigned long long)h);
}

/* per-block histogram of digit at `shift` over the pass input;
 * `tile` elements per block (must match the scatter geometry) */
template <int BIN_MODE>
__global__ void k_radix_hist(int64_t n, const uint64_t* keys, int shift,
                             uint32_t* hist /* [256][nblocks] */, int nblocks,
                             int tile) {
  __shared__ uint32_t h[256];
  if (threadIdx.x < 256) h[threadIdx.x] = 0;
  __syncthreads();
  int64_t base = (int64_t)blockIdx.x * tile;
  int rounds = tile / 256;
  for (int r = 0; r < rounds; r++) {
    int64_t i = base + r * 256 + threadIdx.x;
    if (i < n) atomicAdd(&h[compute_bin<BIN_MODE>(keys[i], shift, 0)], 1u);
  }
  __syncthreads();
  if (threadIdx.x < 256)
    hist[(int64_t)threadIdx.x * nblocks + blockIdx.x] = h[threadIdx.x];
}

/* ---- generic exclusive scan over uint32 (for the hist matrix) ---- */

#define SCAN_BLOCK 256
#define SCAN_ITEMS 16
#define SCAN_TILE (SCAN_BLOCK * SCAN_ITEMS)

DEV uint32_t wave_inclusive_scan(uint32_t v) {
  for (int off = 1; off < WAVE; off <<= 1) {
    uint32_t u = __shfl_up(v, off);
    if ((int)(threadIdx.x & (WAVE - 1)) >= off) v += u;
  }
  return v;
}

/* per-block inclusive scan of a tile; writes tile total to block_sums */
__global__ void k_scan_partial(int64_t n, const uint32_t* in, uint32_t* out,
                               uint32_t* block_sums) {
  __shared__ uint32_t wsum[SCAN_BLOCK / WAVE];
  __shared__ uint32_t carry_s;
  int64_t base = (int64_t)blockIdx.x * SCAN_TILE;
  int wave = threadIdx.x / WAVE, lane = threadIdx.x & (WAVE - 1);
  uint32_t carry = 0;
  for (int r = 0; r < SCAN_ITEMS; r++) {
    int64_t i = base + r * SCAN_BLOCK + threadIdx.x;
    uint32_t v = (i < n) ? in[i] : 0;
    uint32_t s = wave_inclusive_scan(v);
    if (lane == WAVE - 1) wsum[wave] = s;
    __syncthreads();
    if (threadIdx.x == 0) {
      uint32_t acc = 0;
      for (int w = 0; w < SCAN_BLOCK / WAVE; w++) { uint32_t t = wsum[w]; wsum[w] = acc; acc += t; }
      carry_s = acc;
    }
    __syncthreads();
    if (i < n) out[i] = s + wsum[wave] + carry;
    carry += carry_s;
    __syncthreads();
  }
  if (threadIdx.x == 0) block_sums[blockIdx.x] = carry;
}

/* single-block exclusive scan of block_sums (nblocks <= SCAN_TILE * loops) */
__global__ void k_scan_sums(int64_t n, uint32_t* sums) {
  __shared__ uint32_t wsum[SCAN_BLOCK / WAVE];
  __shared__ uint32_t carry_s;
  int wave = threadIdx.x / WAVE, lane = threadIdx.x & (WAVE - 1);
  uint32_t carry = 0;
  for (int64_t base = 0; base < n; base += SCAN_BLOCK) {
    int64_t i = base + threadIdx.x;
    uint32_t v = (i < n) ? sums[i] : 0;
    uint32_t s = wave_inclusive_scan(v);
    if (lane == WAVE - 1) wsum[wave] = s;
    __syncthreads();
    if (threadIdx.x == 0) {
      uint32_t acc = 0;
      for (int w = 0; w < SCAN_BLOCK / WAVE; w++) { uint32_t t = wsum[w]; wsum[w] = acc; acc += t; }
      carry_s = acc;
    }
    __syncthreads();
    if (i < n) sums[i] = s - v + wsum[wave] + carry;  /* exclusive */
    carry += carry_s;
    __syncthreads();
  }
}

/* add scanned block sums back; also converts inclusive->exclusive */
__global__ void k_scan_add(int64_t n, const uint32_t* in, uint32_t* out,
                           const uint32_t* block_sums) {
  int64_t base = (int64_t)blockIdx.x * SCAN_TILE;
  uint32_t add = block_sums[blockIdx.x];
  for (int r = 0; r < SCAN_ITEMS; r++) {
    int64_t i = base + r * SCAN_BLOCK + threadIdx.x;
    if (i < n) out[i] = out[i] - in[i] + add;  /* inclusive - v = exclusive */
  }
}

static int exclusive_scan_u32(hipStream_t s, int64_t n, const uint32_t* in,
                              uint32_t* out, uint32_t* block_sums /* >= nblocks+1 */) {
  int64_t nblocks = (n + SCAN_TILE - 1) / SCAN_TILE;
  k_scan_partial<<<dim3((uint32_t)nblocks), SCAN_BLOCK, 0, s>>>(n, in, out, block_sums);
  k_scan_sums<<<1, SCAN_BLOCK, 0, s>>>(nblocks, block_sums);
  k_scan_add<<<dim3((uint32_t)nblocks), SCAN_BLOCK, 0, s>>>(n, in, out, block_sums);
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

/* status word for decoupled lookback: [63:56] epoch, [55:54] status
 * (1 = aggregate, 2 = inclusive prefix), [53:0] value. One 8-byte
 * agent-scope atomic store/load per word (single-granule publish: the
 * payload IS the flag — no fences needed). */
#define OSW_AGG (1ULL << 54)
#define OSW_PFX (2ULL << 54)
#define OSW_VAL(x) ((x) & ((1ULL << 54) - 1))
#define OSW_EPOCH(x) ((x) >> 56)
#define OSW_SPIN_LIMIT (1u << 22)

typedef unsigned long long ull2_a8 __attribute__((vector_size(16), aligned(8)));
typedef uint32_t u32v2_a4 __attribute__((vector_size(8), aligned(4)));

template <int BIN_MODE, int BLOCK, int ITEMS, bool LOOKBACK, typename PayT = uint32_t,
          int RADIX_BITS = 8, bool CONTIG = false>
__global__ __launch_bounds__(BLOCK)
void k_radix_scatter(int64_t n, const uint64_t* kin, const PayT* iin,
                     uint64_t* kout, PayT* iout,
                     const uint32_t* scanned /* [256][nblocks] exclusive */,
                     int shift, int nblocks, int nparts,
                     unsigned long long* state /* [nblocks][256] */,
                     const uint32_t* gbase /* [256] pass bin bases */,
                     unsigned long long* err_flag, uint32_t epoch,
                     void* decode_out /* non-null on the final pass: write
                       decoded keys here instead of encoded keys to kout */,
                     int decode_mode /* 1=i64 2=f64, +4 = desc */) {
  constexpr int WAVES = BLOCK / WAVE;
  constexpr int TILE = BLOCK * ITEMS;
  constexpr int BINS = 1 << RADIX_BITS;
  __shared__ uint32_t wave_hist[WAVES][BINS];
  __shared__ uint32_t bin_start[BINS];    /* in-block exclusive start per bin */
  __shared__ uint32_t bin_gbase[BINS];    /* global dest minus local start    */
  __shared__ uint32_t wtot[WAVES <= 4 ? 4 : WAVES];
  /* narrow-digit cooperative lookback needs each bin's block count visible
   * to its resolver wave */
  __shared__ uint32_t bin_cnt[(LOOKBACK && RADIX_BITS < 8) ? BINS : 1];
  __shared__ uint64_t stage_k[TILE];
  __shared__ PayT stage_i[TILE];

  const int tid = threadIdx.x;
  const int wave = tid / WAVE, lane = tid & (WAVE - 1);
  const int64_t base = (int64_t)blockIdx.x * TILE;
  const int tile_n = (int)min((int64_t)TILE, n - base);

  for (int b = tid; b < WAVES * BINS; b += BLOCK)
    ((uint32_t*)wave_hist)[b] = 0;
  __syncthreads();

  /* wave w owns the contiguous sub-tile [w*WAVE*ITEMS, ...): element order
   * within the block = (wave, round, lane) = linear tile order. */
  uint64_t k[ITEMS];
  PayT id[ITEMS];
  uint16_t lrank[ITEMS];
  uint8_t lbin[ITEMS];

  const int64_t wbase = base + (int64_t)wave * WAVE * ITEMS;
  /* issue every load before the serial ranking chain so all ITEMS pairs
   * are in flight at once (A/B via GPUQ_SCATTER_NO_PRELOAD) */
  #pragma unroll
  for (int r = 0; r < ITEMS; r++) {
    int64_t i = wbase + r * WAVE + lane;
    bool valid = i < n;
    k[r] = valid ? kin[i] : 0;
    id[r] = valid ? iin[i] : 0;
  }
  for (int r = 0; r < ITEMS; r++) {
    int64_t i = wbase + r * WAVE + lane;
    bool valid = i < n;
    int bin = valid ? compute_bin<BIN_MODE>(k[r], shift, nparts) : 0;
    if (RADIX_BITS < 8) bin &= (1 << RADIX_BITS) - 1;
    lbin[r] = (uint8_t)bin;
    /* ballot multi-split: mask of lanes in this wave with the same bin */
    uint64_t active = __ballot(valid);
    uint64_t same = active;
    for (int b = 0; b < RADIX_BITS; b++) {
      uint64_t bl = __ballot((bin >> b) & 1);
      same &= ((bin >> b) & 1) ? bl : ~bl;
    }
    uint64_t below = same & ((1ULL << lane) - 1);
    int rank = __popcll(below);
    int leader = __ffsll((unsigned long long)same) - 1;
    uint32_t basecnt = 0;
    if (valid && lane == leader) {
      basecnt = wave_hist[wave][bin];
      wave_hist[wave][bin] = basecnt + __popcll(same);
    }
    basecnt = __shfl(basecnt, leader);
    lrank[r] = (uint16_t)(basecnt + rank);
    /* wave_hist rows are wave-private: the round-r leader's LDS RMW only
     * needs to be ordered before round r+1's read within THIS wave. DS ops
     * from one wave are serviced in issue order; the wave barrier stops the
     * compiler from reordering them. */
    __builtin_amdgcn_wave_barrier();
  }
  __syncthreads();

  /* cross-wave exclusive prefix per bin + block-wide exclusive scan over
   * bins. The first 256 threads each own one bin. */
  uint32_t acc = 0;
  if (tid < BINS) {
    int bin = tid;
    for (int w = 0; w < WAVES; w++) {
      uint32_t t = wave_hist[w][bin];
      wave_hist[w][bin] = acc;
      acc += t;
    }
    if (LOOKBACK) {
      /* publish own AGGREGATE early so successors can proceed */
      __hip_atomic_store(&state[(int64_t)blockIdx.x * 256 + bin],
                         ((unsigned long long)epoch << 56) | OSW_AGG | acc,
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      if (RADIX_BITS < 8) bin_cnt[bin] = acc;
    }
    uint32_t inc = wave_inclusive_scan(acc);
    if (lane == WAVE - 1) wtot[wave] = inc;
    bin_start[bin] = inc - acc;  /* provisional; add wave offsets after sync */
  }
  __syncthreads();
  if (tid < BINS) {
    int bin = tid;
    uint32_t woff = 0;
    for (int w = 0; w < wave; w++) woff += wtot[w];
    uint32_t excl = bin_start[bin] + woff;
    bin_start[bin] = excl;
    if (!LOOKBACK) {
      bin_gbase[bin] = scanned[(int64_t)bin * nblocks + blockIdx.x] - excl;
    } else {
      bin_gbase[bin] = excl;  /* stash; lookback resolves after staging */
    }
  }
  __syncthreads();

  /* stage reordered tile in LDS (lookback waits overlap with this) */
  for (int r = 0; r < ITEMS; r++) {
    int64_t i = wbase + r * WAVE + lane;
    if (i < n) {
      uint32_t pos = bin_start[lbin[r]] + wave_hist[wave][lbin[r]] + lrank[r];
      stage_k[pos] = k[r];
      stage_i[pos] = id[r];
    }
  }
  if (LOOKBACK && RADIX_BITS < 8) {
    /* full-wave cooperative lookback: each wave resolves whole bins — all
     * 64 lanes poll one predecessor status word each per round (the
     * per-thread serial walk left only BINS walker lanes active, which
     * measured as the 4-bit mode's wall: 83% WAIT_ANY). Lane j inspects
     * predecessor blockIdx.x-1-j; a ballot finds the newest PREFIX and the
     * first not-ready word, a shuffle reduction sums the consumable run. */
    constexpr int BPW = (BINS + WAVES - 1) / WAVES;
    for (int bi = 0; bi < BPW; bi++) {
      int bin = wave * BPW + bi;
      if (bin >= BINS) break;
      uint32_t excl = bin_gbase[bin];
      unsigned long long pred = 0;
      uint32_t spins = 0;
      int p = (int)blockIdx.x - 1;
      bool failed = false;
      while (p >= 0) {
        int cnt = (p + 1 < WAVE) ? p + 1 : WAVE;
        unsigned long long v = 0;
        bool ready = false, ispfx = false;
        if (lane < cnt) {
          v = __hip_atomic_load(&state[(int64_t)(p - lane) * 256 + bin],
                                __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
          ready = OSW_EPOCH(v) == epoch && (v & (OSW_AGG | OSW_PFX));
          ispfx = ready && (v & OSW_PFX);
        }
        uint64_t nr_mask = __ballot(lane < cnt && !ready);
        uint64_t pfx_mask = __ballot(ispfx);
        int first_nr = nr_mask ? __ffsll((unsigned long long)nr_mask) - 1 : 64;
        int first_pfx = pfx_mask ? __ffsll((unsigned long long)pfx_mask) - 1 : 64;
        bool done = first_pfx < first_nr;
        int take = done ? first_pfx + 1 : (first_nr < cnt ? first_nr : cnt);
        if (take == 0) {
          if (++spins > OSW_SPIN_LIMIT) {
            if (lane == 0)
              __hip_atomic_store(err_flag, 1ull, __ATOMIC_RELAXED,
                                 __HIP_MEMORY_SCOPE_AGENT);
            failed = true;
            break;
          }
          __builtin_amdgcn_s_sleep(1);
          continue;
        }
        unsigned long long contrib = (lane < take) ? OSW_VAL(v) : 0;
        #pragma unroll
        for (int off = WAVE / 2; off > 0; off >>= 1)
          contrib += __shfl_down(contrib, off);
        contrib = __shfl(contrib, 0);
        pred += contrib;
        spins = 0;
        if (done) break;
        p -= take;
      }
      (void)failed;
      if (lane == 0) {
        __hip_atomic_store(&state[(int64_t)blockIdx.x * 256 + bin],
                           ((unsigned long long)epoch << 56) | OSW_PFX |
                               (pred + bin_cnt[bin]),
                           __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        bin_gbase[bin] = gbase[bin] + (uint32_t)pred - excl;
      }
    }
  } else if (LOOKBACK && tid < 256) {
    int bin = tid;
    uint32_t excl = bin_gbase[bin];
    uint32_t own = acc;   /* this thread's phase-2 block count for its bin */
    /* chunked decoupled lookback: LB predecessor words in flight per step */
    constexpr int LB = 4;
    unsigned long long pred = 0;
    uint32_t spins = 0;
    int p = (int)blockIdx.x - 1;
    while (p >= 0) {
      int cnt = (p + 1 < LB) ? p + 1 : LB;
      unsigned long long v[LB];
      #pragma unroll
      for (int j = 0; j < LB; j++)
        if (j < cnt)
          v[j] = __hip_atomic_load(&state[(int64_t)(p - j) * 256 + bin],
                                   __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      /* walk newest -> oldest; stop at first PREFIX; retry at not-ready */
      bool stall = false, done = false;
      unsigned long long add = 0;
      #pragma unroll
      for (int j = 0; j < LB; j++) {
        if (j >= cnt || done || stall) continue;
        if (OSW_EPOCH(v[j]) != epoch || !(v[j] & (OSW_AGG | OSW_PFX))) {
          stall = true;
        } else {
          add += OSW_VAL(v[j]);
          if (v[j] & OSW_PFX) done = true;
        }
      }
      if (stall) {
        if (++spins > OSW_SPIN_LIMIT) {        /* bounded: give up, flag host */
          __hip_atomic_store(err_flag, 1ull, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT);
          break;
        }
        __builtin_amdgcn_s_sleep(1);
        continue;                               /* retry same chunk */
      }
      pred += add;
      if (done) break;
      p -= cnt;
    }
    /* publish inclusive prefix (even after timeout, to unblock others) */
    __hip_atomic_store(&state[(int64_t)blockIdx.x * 256 + bin],
                       ((unsigned long long)epoch << 56) | OSW_PFX |
                           (pred + own),
                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    bin_gbase[bin] = gbase[bin] + (uint32_t)pred - excl;
  }
  __syncthreads();

  /* drain LDS -> coalesced global runs per bin */
  if (CONTIG) {
    /* contiguous per-thread drain: each thread owns ITEMS consecutive
     * staged elements; same-bin neighbours have consecutive destinations,
     * so pairs collapse into ONE 16 B key store + ONE 8 B id store — the
     * same bytes in HALF the store instructions (the padded-record test
     * showed extra BYTES lose; this halves requests at equal bytes).
     * Needs odd ITEMS for tolerable LDS bank conflicts (stride 2*ITEMS
     * words, gcd 2 -> 2-way). PayT must be 4 B. */
    auto out_word = [&](uint64_t kk) -> uint64_t {
      if (!decode_mode) return kk;
      uint64_t e = (decode_mode & 4) ? ~kk : kk;
      if (decode_mode & 2) {
        uint64_t mask = ((e >> 63) ? 0 : ~0ULL) | SIGNBIT;
        return e ^ mask;
      }
      return e ^ SIGNBIT;
    };
    uint64_t* kbase = decode_mode ? (uint64_t*)decode_out : kout;
    int j0 = tid * ITEMS;
    int jend = j0 + ITEMS < tile_n ? j0 + ITEMS : tile_n;
    for (int j = j0; j < jend;) {
      uint64_t k0 = stage_k[j];
      int b0 = compute_bin<BIN_MODE>(k0, shift, nparts);
      if (RADIX_BITS < 8) b0 &= (1 << RADIX_BITS) - 1;
      uint32_t dst = bin_gbase[b0] + (uint32_t)j;
      if (j + 1 < jend) {
        uint64_t k1 = stage_k[j + 1];
        int b1 = compute_bin<BIN_MODE>(k1, shift, nparts);
        if (RADIX_BITS < 8) b1 &= (1 << RADIX_BITS) - 1;
        if (b1 == b0) {
          *(ull2_a8*)&kbase[dst] = (ull2_a8){out_word(k0), out_word(k1)};
          *(u32v2_a4*)&iout[dst] =
              (u32v2_a4){(uint32_t)stage_i[j], (uint32_t)stage_i[j + 1]};
          j += 2;
          continue;
        }
      }
      kbase[dst] = out_word(k0);
      iout[dst] = stage_i[j];
      j += 1;
    }
    return;
  }
  for (int r = 0; r < ITEMS; r++) {
    int j = r * BLOCK + tid;
    if (j < tile_n) {
      uint64_t kk = stage_k[j];
      int bin = compute_bin<BIN_MODE>(kk, shift, nparts);
      if (RADIX_BITS < 8) bin &= (1 << RADIX_BITS) - 1;
      uint32_t dst = bin_gbase[bin] + (uint32_t)j;
      uint64_t kv_out;
      void* kdst;
      if (decode_mode) {
        /* final pass: emit the DECODED key column directly (the separate
           decode kernel and this pass's encoded-key write both disappear) */
        uint64_t e = (decode_mode & 4) ? ~kk : kk;
        if (decode_mode & 2) {
          uint64_t mask = ((e >> 63) ? 0 : ~0ULL) | SIGNBIT;
          kv_out = e ^ mask;
        } else {
          kv_out = e ^ SIGNBIT;
        }
        kdst = &((uint64_t*)decode_out)[dst];
      } else {
        kv_out = kk;
        kdst = &kout[dst];
      }
#ifdef GPUQ_DRAIN_SC1
      /* write-through: scattered partial-line stores skip the L2
       * write-allocate read-for-ownership (guide: publish-large) */
      __hip_atomic_store((uint64_t*)kdst, kv_out, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
      __hip_atomic_store((PayT*)&iout[dst], stage_i[j], __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
#elif defined(GPUQ_DRAIN_NT)
      __builtin_nontemporal_store(kv_out, (uint64_t*)kdst);
      __builtin_nontemporal_store(stage_i[j], &iout[dst]);
#else
      *(uint64_t*)kdst = kv_out;
      iout[dst] = stage_i[j];
#endif
    }
  }
}

/* runtime-selectable geometry (GPUQ_SORT_GEOM env: "BLOCKxITEMS") */
static scatter_geom get_sort_geom(void) {
  static scatter_geom g = {0, 0};
  if (g.block == 0) {
    const char* e = getenv("GPUQ_SORT_GEOM");
    int b = 0, it = 0;
    if (e && sscanf(e, "%dx%d", &b, &it) == 2) { g.block = b; g.items = it; }
    else { g.block = 256; g.items = 22; }  /* two-box A/B winner: ~5% over
                                            * 512x10 (fewer waves per
                                            * barrier at the same tile
                                            * residency) */
    /* only geometries with a dispatch entry are legal: an unknown pair
     * would silently run a different tile than the block count assumed
     * (incomplete sort). Clamp to the default. */
    static const int known[][2] = {{256,16},{512,8},{512,16},{1024,8},{512,4},
                                   {512,6},{256,12},{512,12},{1024,5},{512,10},
                                   {1024,2},{1024,6},{1024,4},{512,11},
                                   {256,20},{256,22}};
    bool ok = false;
    for (auto& k : known) ok = ok || (k[0] == g.block && k[1] == g.items);
    if (!ok) {
      fprintf(stderr, "gpuq: unsupported GPUQ_SORT_GEOM %dx%d, using 256x22\n",
              g.block, g.items);
      g.block = 256; g.items = 22;
    }
  }
  return g;
}

template <int BIN_MODE, bool LOOKBACK>
static void launch_scatter(hipStream_t s, scatter_geom g, int64_t nb,
                           int64_t n, const uint64_t* kin, const uint32_t* iin,
                           uint64_t* kout, uint32_t* iout,
                           const uint32_t* scanned, int shift, int nparts,
                           unsigned long long* state, const uint32_t* gbase,
                           unsigned long long* err_flag, uint32_t epoch,
                           void* decode_out = nullptr, int decode_mode = 0) {
  dim3 grid((uint32_t)nb);
#define LS(B, I) k_radix_scatter<BIN_MODE, B, I, LOOKBACK><<<grid, B, 0, s>>>( \
      n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase, \
      err_flag, epoch, decode_out, decode_mode)
  static int contig = -1;
  if (contig < 0) contig = getenv("GPUQ_DRAIN_CONTIG") != nullptr;
  if (contig && g.block == 512 && (g.items == 9 || g.items == 10 || g.items == 11)) {
    if (g.items == 9)
      k_radix_scatter<BIN_MODE, 512, 9, LOOKBACK, uint32_t, 8, true><<<grid, 512, 0, s>>>(
          n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase,
          err_flag, epoch, decode_out, decode_mode);
    else if (g.items == 11)
      k_radix_scatter<BIN_MODE, 512, 11, LOOKBACK, uint32_t, 8, true><<<grid, 512, 0, s>>>(
          n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase,
          err_flag, epoch, decode_out, decode_mode);
    else
      k_radix_scatter<BIN_MODE, 512, 10, LOOKBACK, uint32_t, 8, true><<<grid, 512, 0, s>>>(
          n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase,
          err_flag, epoch, decode_out, decode_mode);
    return;
  }
  if (g.block == 256 && g.items == 20)
    k_radix_scatter<BIN_MODE, 256, 20, LOOKBACK><<<grid, 256, 0, s>>>(
      n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase,
      err_flag, epoch, decode_out, decode_mode);
  else if (g.block == 256 && g.items == 22)
    k_radix_scatter<BIN_MODE, 256, 22, LOOKBACK><<<grid, 256, 0, s>>>(
      n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase,
      err_flag, epoch, decode_out, decode_mode);
  else if (g.block == 512 && g.items == 11)
    k_radix_scatter<BIN_MODE, 512, 11, LOOKBACK><<<grid, 512, 0, s>>>(
      n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase,
      err_flag, epoch, decode_out, decode_mode);
  else if (g.block == 256 && g.items == 16) LS(256, 16);
  else if (g.block == 512 && g.items == 8) LS(512, 8);
  else if (g.block == 512 && g.items == 16) LS(512, 16);
  else if (g.block == 1024 && g.items == 8) LS(1024, 8);
  else if (g.block == 512 && g.items == 4) LS(512, 4);
  else if (g.block == 512 && g.items == 6)
    k_radix_scatter<BIN_MODE, 512, 6, LOOKBACK><<<grid, 512, 0, s>>>(
      n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase,
      err_flag, epoch, decode_out, decode_mode);
  else if (g.block == 256 && g.items == 12)
    k_radix_scatter<BIN_MODE, 256, 12, LOOKBACK><<<grid, 256, 0, s>>>(
      n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase,
      err_flag, epoch, decode_out, decode_mode);
  else if (g.block == 512 && g.items == 12)
    k_radix_scatter<BIN_MODE, 512, 12, LOOKBACK><<<grid, 512, 0, s>>>(
      n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase,
      err_flag, epoch, decode_out, decode_mode);
  else if (g.block == 1024 && g.items == 5)
    k_radix_scatter<BIN_MODE, 1024, 5, LOOKBACK><<<grid, 1024, 0, s>>>(
      n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase,
      err_flag, epoch, decode_out, decode_mode);
  else if (g.block == 512 && g.items == 10)
    k_radix_scatter<BIN_MODE, 512, 10, LOOKBACK><<<grid, 512, 0, s>>>(
      n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase,
      err_flag, epoch, decode_out, decode_mode);
  else if (g.block == 1024 && g.items == 2) LS(1024, 2);
  else if (g.block == 1024 && g.items == 6) LS(1024, 6);
  else LS(1024, 4);
#undef LS
}

/* ---- scatter phase ablation (diagnostic only; tools/diag_scatter.py) ----
 * MODE 0: loads only (checksummed against DCE)  1: +ranking  2: +staging
 * 3: +drain (= the real kernel's work). Geometry fixed 512x10. */
template <int MODE>
__global__ __launch_bounds__(512)
void k_scatter_ablate(int64_t n, const uint64_t* kin, const uint32_t* iin,
                      uint64_t* kout, uint32_t* iout,
                      const uint32_t* gbase, unsigned long long* sink,
                      int shift) {
  constexpr int BLOCK = 512, ITEMS = 10, WAVES = BLOCK / WAVE;
  constexpr int TILE = BLOCK * ITEMS;
  __shared__ uint32_t wave_hist[WAVES][256];
  __shared__ uint32_t bin_start[256];
  __shared__ uint32_t bin_gbase[256];
  __shared__ uint32_t wtot[WAVES < 4 ? 4 : WAVES];
  __shared__ uint64_t stage_k[TILE];
  __shared__ uint32_t stage_i[TILE];
  const int tid = threadIdx.x;
  const int wave = tid / WAVE, lane = tid & (WAVE - 1);
  const int64_t base = (int64_t)blockIdx.x * TILE;
  const int tile_n = (int)min((int64_t)TILE, n - base);
  for (int b = tid; b < WAVES * 256; b += BLOCK) ((uint32_t*)wave_hist)[b] = 0;
  __syncthreads();
  uint64_t k[ITEMS]; uint32_t id[ITEMS]; uint16_t lrank[ITEMS]; uint8_t lbin[ITEMS];
  const int64_t wbase = base + (int64_t)wave * WAVE * ITEMS;
  #pragma unroll
  for (int r = 0; r < ITEMS; r++) {
    int64_t i = wbase + r * WAVE + lane;
    bool valid = i < n;
    k[r] = valid ? kin[i] : 0;
    id[r] = valid ? iin[i] : 0;
  }
  if (MODE == 0) {
    uint64_t acc = 0;
    for (int r = 0; r < ITEMS; r++) acc ^= k[r] + id[r];
    if (acc == 0xDEADBEEFCAFEBABEull) atomicAdd(sink, 1ull);  /* keep loads */
    return;
  }
  for (int r = 0; r < ITEMS; r++) {
    int64_t i = wbase + r * WAVE + lane;
    bool valid = i < n;
    int bin = valid ? (int)((k[r] >> shift) & 0xff) : 0;
    lbin[r] = (uint8_t)bin;
    uint64_t m = __ballot(valid);
    for (int b = 0; b < 8; b++) {
      uint64_t bl = __ballot((bin >> b) & 1);
      m &= ((bin >> b) & 1) ? bl : ~bl;
    }
    uint64_t below = m & ((1ULL << lane) - 1);
    int rank = __popcll(below);
    int leader = __ffsll((unsigned long long)m) - 1;
    uint32_t basecnt = 0;
    if (valid && lane == leader) {
      basecnt = wave_hist[wave][bin];
      wave_hist[wave][bin] = basecnt + __popcll(m);
    }
    basecnt = __shfl(basecnt, leader);
    lrank[r] = (uint16_t)(basecnt + rank);
    __builtin_amdgcn_wave_barrier();
  }
  __syncthreads();
  uint32_t acc = 0;
  if (tid < 256) {
    int bin = tid;
    for (int w = 0; w < WAVES; w++) {
      uint32_t t = wave_hist[w][bin];
      wave_hist[w][bin] = acc; acc += t;
    }
    uint32_t inc = wave_inclusive_scan(acc);
    if (lane == WAVE - 1) wtot[wave] = inc;
    bin_start[bin] = inc - acc;
  }
  __syncthreads();
  if (tid < 256) {
    int bin = tid;
    uint32_t woff = 0;
    for (int w = 0; w < wave; w++) woff += wtot[w];
    uint32_t excl = bin_start[bin] + woff;
    bin_start[bin] = excl;
    /* realistic layout: uniform data contributes ~TILE/256 rows per bin per
     * block, so block b's bin-k run starts near gbase[k] + b*TILE/256 —
     * the same write pattern the real scanned offsets produce */
    bin_gbase[bin] = gbase[bin] + (uint32_t)(base >> 8) - excl;
  }
  __syncthreads();
  if (MODE == 1) {
    uint32_t a2 = 0;
    for (int r = 0; r < ITEMS; r++) a2 += lrank[r] + bin_start[lbin[r]];
    if (a2 == 0xDEADBEEFu) atomicAdd(sink, 1ull);
    return;
  }
  for (int r = 0; r < ITEMS; r++) {
    int64_t i = wbase + r * WAVE + lane;
    if (i < n) {
      uint32_t pos = bin_start[lbin[r]] + wave_hist[wave][lbin[r]] + lrank[r];
      stage_k[pos] = k[r];
      stage_i[pos] = id[r];
    }
  }
  __syncthreads();
  if (MODE == 2) {
    uint64_t a3 = stage_k[tid] + stage_i[TILE - 1 - tid];
    if (a3 == 0xDEADBEEFCAFEBABEull) atomicAdd(sink, 1ull);
    return;
  }
  for (int r = 0; r < ITEMS; r++) {
    int j = r * BLOCK + tid;
    if (j < tile_n) {
      uint64_t kk = stage_k[j];
      int bin = (int)((kk >> shift) & 0xff);
      uint32_t dst = (bin_gbase[bin] + (uint32_t)j) % (uint32_t)n;
      kout[dst] = kk;
      iout[dst] = stage_i[j];
    }
  }
}

/* MODE 4/5: 512x8 tile-4096 variant comparing drain store shapes:
 * 4 = two stores per element (8B key + 4B idx, separate arrays, as shipped)
 * 5 = one 16B record store per element (key, idx, pad) */
template <int MODE>
__global__ __launch_bounds__(512)
void k_scatter_ablate2(int64_t n, const uint64_t* kin, const uint32_t* iin,
                       uint64_t* kout, uint32_t* iout, uint32_t* rout,
                       const uint32_t* gbase, int shift) {
  constexpr int BLOCK = 512, ITEMS = 8;
  constexpr int WAVES = BLOCK / WAVE;
  constexpr int TILE = BLOCK * ITEMS;
  __shared__ uint32_t wave_hist[WAVES][256];
  __shared__ uint32_t bin_start[256];
  __shared__ uint32_t bin_gbase[256];
  __shared__ uint32_t wtot[WAVES];
  __shared__ uint64_t stage_k[TILE];
  __shared__ uint32_t stage_i[TILE];
  const int tid = threadIdx.x;
  const int wave = tid / WAVE, lane = tid & (WAVE - 1);
  const int64_t base = (int64_t)blockIdx.x * TILE;
  const int tile_n = (int)min((int64_t)TILE, n - base);
  for (int b = tid; b < WAVES * 256; b += BLOCK) ((uint32_t*)wave_hist)[b] = 0;
  __syncthreads();
  uint64_t k[ITEMS]; uint32_t id[ITEMS]; uint16_t lrank[ITEMS]; uint8_t lbin[ITEMS];
  const int64_t wbase = base + (int64_t)wave * WAVE * ITEMS;
  #pragma unroll
  for (int r = 0; r < ITEMS; r++) {
    int64_t i = wbase + r * WAVE + lane;
    bool valid = i < n;
    k[r] = valid ? kin[i] : 0;
    id[r] = valid ? iin[i] : 0;
  }
  for (int r = 0; r < ITEMS; r++) {
    int64_t i = wbase + r * WAVE + lane;
    bool valid = i < n;
    int bin = valid ? (int)((k[r] >> shift) & 0xff) : 0;
    lbin[r] = (uint8_t)bin;
    uint64_t m = __ballot(valid);
    for (int b = 0; b < 8; b++) {
      uint64_t bl = __ballot((bin >> b) & 1);
      m &= ((bin >> b) & 1) ? bl : ~bl;
    }
    uint64_t below = m & ((1ULL << lane) - 1);
    int rank = __popcll(below);
    int leader = __ffsll((unsigned long long)m) - 1;
    uint32_t basecnt = 0;
    if (valid && lane == leader) {
      basecnt = wave_hist[wave][bin];
      wave_hist[wave][bin] = basecnt + __popcll(m);
    }
    basecnt = __shfl(basecnt, leader);
    lrank[r] = (uint16_t)(basecnt + rank);
    __builtin_amdgcn_wave_barrier();
  }
  __syncthreads();
  if (tid < 256) {
    int bin = tid;
    uint32_t acc = 0;
    for (int w = 0; w < WAVES; w++) {
      uint32_t t = wave_hist[w][bin];
      wave_hist[w][bin] = acc; acc += t;
    }
    uint32_t inc = wave_inclusive_scan(acc);
    if (lane == WAVE - 1) wtot[wave] = inc;
    bin_start[bin] = inc - acc;
  }
  __syncthreads();
  if (tid < 256) {
    int bin = tid;
    uint32_t woff = 0;
    for (int w = 0; w < wave; w++) woff += wtot[w];
    uint32_t excl = bin_start[bin] + woff;
    bin_start[bin] = excl;
    bin_gbase[bin] = gbase[bin] + (uint32_t)(base >> 8) - excl;
  }
  __syncthreads();
  for (int r = 0; r < ITEMS; r++) {
    int64_t i = wbase + r * WAVE + lane;
    if (i < n) {
      uint32_t pos = bin_start[lbin[r]] + wave_hist[wave][lbin[r]] + lrank[r];
      stage_k[pos] = k[r];
      stage_i[pos] = id[r];
    }
  }
  __syncthreads();
  for (int r = 0; r < ITEMS; r++) {
    int j = r * BLOCK + tid;
    if (j < tile_n) {
      uint64_t kk = stage_k[j];
      int bin = (int)((kk >> shift) & 0xff);
      uint32_t dst = (bin_gbase[bin] + (uint32_t)j) % (uint32_t)n;
      if (MODE == 4) {
        kout[dst] = kk;
        iout[dst] = stage_i[j];
      } else {
        uint4 rec;
        rec.x = (uint32_t)kk; rec.y = (uint32_t)(kk >> 32);
        rec.z = stage_i[j]; rec.w = 0;
        ((uint4*)rout)[dst] = rec;
      }
    }
  }
}

extern "C" int gpuq_scatter_ablate2(void* stream, int64_t n, const void* kin,
                                    const void* iin, void* kout, void* iout,
                                    void* rout, const void* gbase, int32_t mode) {
  hipStream_t s = (hipStream_t)stream;
  int64_t nb = (n + 4095) / 4096;
  if (mode == 4)
    k_scatter_ablate2<4><<<dim3((uint32_t)nb), 512, 0, s>>>(
        n, (const uint64_t*)kin, (const uint32_t*)iin, (uint64_t*)kout,
        (uint32_t*)iout, (uint32_t*)rout, (const uint32_t*)gbase, 0);
  else
    k_scatter_ablate2<5><<<dim3((uint32_t)nb), 512, 0, s>>>(
        n, (const uint64_t*)kin, (const uint32_t*)iin, (uint64_t*)kout,
        (uint32_t*)iout, (uint32_t*)rout, (const uint32_t*)gbase, 0);
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

extern "C" int gpuq_scatter_ablate(void* stream, int64_t n, const void* kin,
                                   const void* iin, void* kout, void* iout,
                                   const void* gbase, void* sink, int32_t mode) {
  hipStream_t s = (hipStream_t)stream;
  int64_t nb = (n + 5119) / 5120;
  dim3 g((uint32_t)nb);
#define AB(M) k_scatter_ablate<M><<<g, 512, 0, s>>>(n, (const uint64_t*)kin, \
    (const uint32_t*)iin, (uint64_t*)kout, (uint32_t*)iout, \
    (const uint32_t*)gbase, (unsigned long long*)sink, 0)
  if (mode == 0) AB(0); else if (mode == 1) AB(1);
  else if (mode == 2) AB(2); else AB(3);
#undef AB
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

template <int BIN_MODE, bool LOOKBACK>
static void launch_scatter4(hipStream_t s, scatter_geom g, int64_t nb,
                            int64_t n, const uint64_t* kin, const uint32_t* iin,
                            uint64_t* kout, uint32_t* iout,
                            const uint32_t* scanned, int shift, int nparts,
                            unsigned long long* state, const uint32_t* gbase,
                            unsigned long long* err_flag, uint32_t epoch,
                            void* decode_out = nullptr, int decode_mode = 0) {
  dim3 grid((uint32_t)nb);
  if (g.block == 1024 && g.items == 8)
    k_radix_scatter<BIN_MODE, 1024, 8, LOOKBACK, uint32_t, 4><<<grid, 1024, 0, s>>>(
        n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase,
        err_flag, epoch, decode_out, decode_mode);
  else if (g.block == 256 && g.items == 22)
    k_radix_scatter<BIN_MODE, 256, 22, LOOKBACK, uint32_t, 4><<<grid, 256, 0, s>>>(
        n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase,
        err_flag, epoch, decode_out, decode_mode);
  else if (g.block == 512 && g.items == 10)
    k_radix_scatter<BIN_MODE, 512, 10, LOOKBACK, uint32_t, 4><<<grid, 512, 0, s>>>(
        n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase,
        err_flag, epoch, decode_out, decode_mode);
  else
    k_radix_scatter<BIN_MODE, 512, 10, LOOKBACK, uint32_t, 4><<<grid, 512, 0, s>>>(
        n, kin, iin, kout, iout, scanned, shift, (int)nb, nparts, state, gbase,
        err_flag, epoch, decode_out, decode_mode);  /* unreachable: gated above */
}

/* decode sorted keys back to the output dtype */
template <int DTYPE, bool DESC>
__global__ void k_decode(int64_t n, const uint64_t* ek, void* out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    uint64_t e = ek[i];
    if (DESC) e = ~e;
    if (DTYPE == GPUQ_FLOAT64) {
      uint64_t mask = ((e >> 63) ? 0 : ~0ULL) | SIGNBIT;  /* inverse bijection */
      ((uint64_t*)out)[i] = e ^ mask;
    } else {
      ((int64_t*)out)[i] = (int64_t)(e ^ SIGNBIT);
    }
  }
}

/* workspace layout for sort/partition (geometry-aware: the GPUQ_SORT_GEOM
 * env must not change between workspace sizing and the sort call) */
struct sort_ws {
  uint64_t* ka; uint64_t* kb;
  uint32_t* ia; uint32_t* ib;
  uint32_t* hist; uint32_t* hist_scan; uint32_t* block_sums;
  unsigned long long* bits; /* [0]=and [1]=or */
  uint32_t* ghist;          /* [8][256] global byte histograms */
  uint32_t* gbase;          /* [8][256] per-pass exclusive bin bases */
  unsigned long long* state; /* [nblocks][256] lookback status words */
  unsigned long long* err;   /* lookback timeout flag */
};

static int64_t sort_nblocks(int64_t n, int tile) { return (n + tile - 1) / tile; }

static void sort_ws_layout(int64_t n, int nbins, sort_ws* w, char* basep, int64_t* total) {
  int tile = get_sort_geom().block * get_sort_geom().items;
  int64_t nb = sort_nblocks(n, tile);
  int64_t hist_n = (int64_t)nbins * nb;
  int64_t scan_blocks = (hist_n + SCAN_TILE - 1) / SCAN_TILE + 1;
  int64_t off = 0;
  auto take = [&](int64_t bytes) {
    char* p = basep ? basep + off : nullptr;
    off += (bytes + 255) & ~255LL;
    return p;
  };
  w->ka = (uint64_t*)take(n * 8);
  w->kb = (uint64_t*)take(n * 8);
  w->ia = (uint32_t*)take(n * 4);
  w->ib = (uint32_t*)take(n * 4);
  w->hist = (uint32_t*)take(hist_n * 4);
  w->hist_scan = (uint32_t*)take(hist_n * 4);
  w->block_sums = (uint32_t*)take(scan_blocks * 4);
  w->bits = (unsigned long long*)take(16);
  w->ghist = (uint32_t*)take(8 * 256 * 4);
  w->gbase = (uint32_t*)take(8 * 256 * 4);
  w->state = (unsigned long long*)take(nb * 256 * 8);
  w->err = (unsigned long long*)take(8);
  *total = off;
}

extern "C" int64_t gpuq_sort_workspace_bytes(int64_t n) {
  sort_ws w; int64_t total;
  sort_ws_layout(n, 256, &w, nullptr, &total);
  return total;
}

static void launch_encode(hipStream_t s, int64_t n, const void* keys,
                          const uint32_t* rowmap, int dtype, int desc, sort_ws* w) {
  if (dtype == GPUQ_FLOAT64) {
    if (desc) k_encode<GPUQ_FLOAT64, true><<<grid1d(n), 256, 0, s>>>(n, keys, rowmap, w->ka, w->ia, &w->bits[0], &w->bits[1], w->ghist);
    else      k_encode<GPUQ_FLOAT64, false><<<grid1d(n), 256, 0, s>>>(n, keys, rowmap, w->ka, w->ia, &w->bits[0], &w->bits[1], w->ghist);
  } else {
    if (desc) k_encode<GPUQ_INT64, true><<<grid1d(n), 256, 0, s>>>(n, keys, rowmap, w->ka, w->ia, &w->bits[0], &w->bits[1], w->ghist);
    else      k_encode<GPUQ_INT64, false><<<grid1d(n), 256, 0, s>>>(n, keys, rowmap, w->ka, w->ia, &w->bits[0], &w->bits[1], w->ghist);
  }
}

extern "C" int gpuq_sort_perm(void* stream, int64_t n, gpuq_col key,
                              int32_t desc, int32_t nulls_first,
                              uint32_t* out_perm, void* out_keys,
                              void* workspace, int64_t workspace_bytes) {
  hipStream_t s = (hipStream_t)stream;
  if (n > 0xFFFFFFFFLL) FAIL(GPUQ_ERR_INVALID, "sort: nrows %lld > 2^32", (long long)n);
  if (key.dtype != GPUQ_INT64 && key.dtype != GPUQ_FLOAT64)
    FAIL(GPUQ_ERR_INVALID, "sort: unsupported dtype %d", key.dtype);
  sort_ws w; int64_t need;
  sort_ws_layout(n, 256, &w, (char*)workspace, &need);
  if (workspace_bytes < need)
    FAIL(GPUQ_ERR_INVALID, "sort: workspace %lld < %lld", (long long)workspace_bytes, (long long)need);
  if (n == 0) return GPUQ_OK;

  scatter_geom geom = get_sort_geom();
  int tile = geom.block * geom.items;
  static __thread int force_classic = 0;
  bool onesweep = !force_classic && getenv("GPUQ_NO_ONESWEEP") == nullptr;

  /* NULL keys: stable split into [nulls][valids] (or the reverse) per
   * nullOrdering (SortOrder.scala:35-45), then radix-sort the valid subset
   * through a row map; the final permutation is the concatenation. */
  int64_t n_sort = n;           /* rows entering the radix passes */
  int64_t sorted_at = 0;        /* offset of the sorted segment in out_perm */
  const uint32_t* rowmap = nullptr;
  if (key.validity) {
    int64_t nbv = sort_nblocks(n, tile);
    HIP_TRY(hipMemsetAsync(w.err, 0, 8, s));
    k_validity_pids<<<grid1d(n), 256, 0, s>>>(n, key.validity, nulls_first,
                                              w.ka, w.ia, w.err);
    HIP_TRY(hipGetLastError());
    k_radix_hist<0><<<dim3((uint32_t)nbv), 256, 0, s>>>(n, w.ka, 0, w.hist, (int)nbv, tile);
    HIP_TRY(hipGetLastError());
    int rc = exclusive_scan_u32(s, 256 * nbv, w.hist, w.hist_scan, w.block_sums);
    if (rc) return rc;
    launch_scatter<0, false>(s, geom, nbv, n, w.ka, w.ia, w.kb, w.ib, w.hist_scan,
                             0, 0, nullptr, nullptr, nullptr, 0);
    HIP_TRY(hipGetLastError());
    unsigned long long hnull = 0;
    HIP_TRY(hipMemcpyAsync(&hnull, w.err, 8, hipMemcpyDeviceToHost, s));
    HIP_TRY(hipStreamSynchronize(s));
    int64_t nnull = (int64_t)hnull;
    n_sort = n - nnull;
    sorted_at = nulls_first ? nnull : 0;
    int64_t nulls_at_out = nulls_first ? 0 : n_sort;      /* in out_perm */
    int64_t nulls_at_ib = nulls_first ? 0 : n_sort;       /* in w.ib (same order) */
    if (nnull > 0)
      HIP_TRY(hipMemcpyAsync(out_perm + nulls_at_out, w.ib + nulls_at_ib,
                             nnull * 4, hipMemcpyDeviceToDevice, s));
    rowmap = w.ib + sorted_at;   /* valid rows, input order; consumed by encode */
    if (n_sort == 0) {
      if (out_keys) {
        gpuq_col kc = key; kc.validity = nullptr;
        return gpuq_gather(stream, n, kc, out_perm, out_keys);
      }
      return GPUQ_OK;
    }
  }

  int64_t nb = sort_nblocks(n_sort, tile);
  HIP_TRY(hipMemsetAsync(w.bits, 0, 16, s));
  HIP_TRY(hipMemsetAsync(w.bits, 0xFF, 8, s));  /* bits_and = ~0 */
  HIP_TRY(hipMemsetAsync(w.ghist, 0, 8 * 256 * 4, s));
  /* encode + skip-byte reduction + all-byte global histograms (one read) */
  { hipEvent_t _pe = prof_begin(s);
  launch_encode(s, n_sort, key.data, rowmap, key.dtype, desc, &w);
  prof_end("encode", s, _pe); }
  HIP_TRY(hipGetLastError());
  unsigned long long hb[2];
  uint32_t hghist[8 * 256];
  HIP_TRY(hipMemcpyAsync(hb, w.bits, 16, hipMemcpyDeviceToHost, s));
  HIP_TRY(hipMemcpyAsync(hghist, w.ghist, sizeof(hghist), hipMemcpyDeviceToHost, s));
  HIP_TRY(hipStreamSynchronize(s));
  uint64_t bits_changed = hb[0] ^ hb[1];

  bool nibble_mode = onesweep && getenv("GPUQ_SORT_BITS") != nullptr &&
                     atoi(getenv("GPUQ_SORT_BITS")) == 4;
  /* the 4-bit kernel templates exist for these geometries only; a tile
   * mismatch would overlap blocks (wrong results) — fall back to 8-bit */
  if (nibble_mode && !((geom.block == 1024 && geom.items == 8) ||
                       (geom.block == 512 && geom.items == 10) ||
                       (geom.block == 256 && geom.items == 22)))
    nibble_mode = false;
  int retries = 0;
retry:
  if (onesweep && !nibble_mode) {
    /* per-pass exclusive bin bases from the one-read global histograms
     * (replaces the per-pass hist kernel + device scan) */
    uint32_t hgbase[8 * 256];
    for (int b = 0; b < 8; b++) {
      uint32_t run = 0;
      for (int bin = 0; bin < 256; bin++) {
        hgbase[b * 256 + bin] = run;
        run += hghist[b * 256 + bin];
      }
    }
    HIP_TRY(hipMemcpyAsync(w.gbase, hgbase, sizeof(hgbase), hipMemcpyHostToDevice, s));
    HIP_TRY(hipMemsetAsync(w.state, 0, nb * 256 * 8, s));
    HIP_TRY(hipMemsetAsync(w.err, 0, 8, s));
  } else if (nibble_mode) {
    /* 16 levels of 4-bit digits (16-bin drains write 16x-longer runs);
     * level bases derived from the same byte histograms */
    uint32_t hgbase[16 * 16];
    for (int lvl = 0; lvl < 16; lvl++) {
      int byte = lvl / 2;
      bool high = lvl & 1;
      uint32_t cnt[16] = {0};
      for (int v = 0; v < 256; v++) {
        int nib = high ? (v >> 4) : (v & 15);
        cnt[nib] += hghist[byte * 256 + v];
      }
      uint32_t run = 0;
      for (int v = 0; v < 16; v++) { hgbase[lvl * 16 + v] = run; run += cnt[v]; }
    }
    HIP_TRY(hipMemcpyAsync(w.gbase, hgbase, sizeof(hgbase), hipMemcpyHostToDevice, s));
    HIP_TRY(hipMemsetAsync(w.state, 0, nb * 256 * 8, s));
    HIP_TRY(hipMemsetAsync(w.err, 0, 8, s));
  }

  /* last active pass writes row ids straight into out_perm (saves the
   * 4 B/row device copy) */
  int nlevels = nibble_mode ? 16 : 8;
  int lvl_bits = nibble_mode ? 4 : 8;
  uint64_t lvl_mask = nibble_mode ? 0xfULL : 0xffULL;
  int last_byte = -1;
  for (int lvl = 0; lvl < nlevels; lvl++)
    if (((bits_changed >> (lvl * lvl_bits)) & lvl_mask) != 0) last_byte = lvl;
  uint64_t *kin = w.ka, *kout = w.kb;
  uint32_t *iin = w.ia, *iout = w.ib;
  for (int byte = 0; byte < nlevels; byte++) {
    if (((bits_changed >> (byte * lvl_bits)) & lvl_mask) == 0) continue;  /* RadixSort.java:126 skip */
    int shift = byte * lvl_bits;
    uint32_t* iout_pass = (byte == last_byte) ? out_perm + sorted_at : iout;
    /* fuse the key decode into the final pass when the caller wants keys
     * and the null path is not rerouting them through a gather */
    void* dec_out = nullptr;
    int dec_mode = 0;
    if (byte == last_byte && out_keys && !key.validity) {
      dec_out = (char*)out_keys;  /* sorted_at == 0 without validity */
      dec_mode = (key.dtype == GPUQ_FLOAT64 ? 2 : 1) | (desc ? 4 : 0);
    }
    if (nibble_mode) {
      { hipEvent_t _pe = prof_begin(s);
      launch_scatter4<0, true>(s, geom, nb, n_sort, kin, iin, kout, iout_pass, nullptr,
                               shift, 0, w.state, w.gbase + byte * 16, w.err,
                               (uint32_t)(byte + 1), dec_out, dec_mode);
      prof_end("radix_scatter", s, _pe); }
      HIP_TRY(hipGetLastError());
    } else if (onesweep) {
      { hipEvent_t _pe = prof_begin(s);
      launch_scatter<0, true>(s, geom, nb, n_sort, kin, iin, kout, iout_pass, nullptr,
                              shift, 0, w.state, w.gbase + byte * 256, w.err,
                              (uint32_t)(byte + 1), dec_out, dec_mode);
      prof_end("radix_scatter", s, _pe); }
      HIP_TRY(hipGetLastError());
    } else {
      { hipEvent_t _pe = prof_begin(s);
      k_radix_hist<0><<<dim3((uint32_t)nb), 256, 0, s>>>(n_sort, kin, shift, w.hist, (int)nb, tile);
      prof_end("radix_hist", s, _pe); }
      HIP_TRY(hipGetLastError());
      int rc = exclusive_scan_u32(s, (int64_t)256 * nb, w.hist, w.hist_scan, w.block_sums);
      if (rc) return rc;
      { hipEvent_t _pe = prof_begin(s);
      launch_scatter<0, false>(s, geom, nb, n_sort, kin, iin, kout, iout_pass, w.hist_scan,
                               shift, 0, nullptr, nullptr, nullptr, 0, dec_out, dec_mode);
      prof_end("radix_scatter", s, _pe); }
      HIP_TRY(hipGetLastError());
    }
    uint64_t* tk = kin; kin = kout; kout = tk;
    uint32_t* ti = iin; iin = iout_pass; iout = ti;
  }

  if (onesweep) {
    /* one bounded-spin timeout check; on timeout redo with hist+scan (the
     * input was only read; pass buffers are scratch) */
    unsigned long long herr = 0;
    HIP_TRY(hipMemcpyAsync(&herr, w.err, 8, hipMemcpyDeviceToHost, s));
    HIP_TRY(hipStreamSynchronize(s));
    if (herr) {
      if (++retries > 1) FAIL(GPUQ_ERR_HIP, "sort: lookback timed out twice");
      onesweep = false;
      /* re-encode (ka/ia were consumed as ping-pong scratch; with a validity
       * split the rowmap in w.ib was also consumed -> rebuild is only needed
       * for the null-free path; with validity, fall back to a full redo) */
      if (key.validity) {
        force_classic = 1;
        int rc = gpuq_sort_perm(stream, n, key, desc, nulls_first, out_perm,
                                out_keys, workspace, workspace_bytes);
        force_classic = 0;
        return rc;
      }
      launch_encode(s, n_sort, key.data, nullptr, key.dtype, desc, &w);
      HIP_TRY(hipGetLastError());
      goto retry;
    }
  }
  if (last_byte < 0)  /* zero passes: identity permutation */
    HIP_TRY(hipMemcpyAsync(out_perm + sorted_at, iin, n_sort * 4, hipMemcpyDeviceToDevice, s));
  if (out_keys) {
    if (key.validity) {
      /* null rows' data values are whatever the input held — gather the
       * whole column by the final permutation */
      gpuq_col kc = key; kc.validity = nullptr;
      int rc = gpuq_gather(stream, n, kc, out_perm, out_keys);
      if (rc) return rc;
    } else if (last_byte >= 0) {
      /* keys were decoded by the fused final pass */
    } else {
      if (key.dtype == GPUQ_FLOAT64) {
        if (desc) k_decode<GPUQ_FLOAT64, true><<<grid1d(n), 256, 0, s>>>(n, kin, out_keys);
        else      k_decode<GPUQ_FLOAT64, false><<<grid1d(n), 256, 0, s>>>(n, kin, out_keys);
      } else {
        if (desc) k_decode<GPUQ_INT64, true><<<grid1d(n), 256, 0, s>>>(n, kin, out_keys);
        else      k_decode<GPUQ_INT64, false><<<grid1d(n), 256, 0, s>>>(n, kin, out_keys);
      }
      HIP_TRY(hipGetLastError());
    }
  }
  return GPUQ_OK;
}

/* ---- partition: one ranked-scatter pass with bin = partition id ---- */

/* pack (validity, key) into the u64 slot used by compute_bin<1>:
 * top bit set = NULL key; low 63 bits = key (keys needing bit 63 are rare
 * in partition keys? NO — must be exact: instead we pre-compute pids). */

/* For exactness with full-range int64 keys we precompute the pid array and
 * use BIN_MODE 0 over its low byte (num_parts <= 256). */
__global__ void k_partition_pids(int64_t n, const int64_t* keys, const uint8_t* validity,
                                 int32_t nparts, uint64_t* pid_as_key, uint32_t* idx,
                                 unsigned long long* counts /* [nparts] */) {
  __shared__ uint32_t h[256];
  bool lds_counts = nparts <= 256;
  if (lds_counts)
    for (int b = threadIdx.x; b < nparts; b += blockDim.x) h[b] = 0;
  __syncthreads();
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int32_t hsh = 42;
    if (bit_valid(validity, i)) hsh = mm3_hash_long(keys[i], 42);
    int pid = spark_pmod(hsh, nparts);
    pid_as_key[i] = (uint64_t)pid;
    idx[i] = (uint32_t)i;
    if (lds_counts) atomicAdd(&h[pid], 1u);
    else atomicAdd(&counts[pid], 1ull);
  }
  __syncthreads();
  if (lds_counts)
    for (int b = threadIdx.x; b < nparts; b += blockDim.x)
      if (h[b]) atomicAdd(&counts[b], (unsigned long long)h[b]);
}

extern "C" int64_t gpuq_partition_workspace_bytes(int64_t n, int32_t nparts) {
  (void)nparts;
  sort_ws w; int64_t total;
  sort_ws_layout(n, 256, &w, nullptr, &total);
  return total;
}

extern "C" int gpuq_partition_perm(void* stream, int64_t n, gpuq_col key,
                                   int32_t nparts, uint32_t* out_perm,
                                   int64_t* out_counts,
                                   void* workspace, int64_t workspace_bytes) {
  hipStream_t s = (hipStream_t)stream;
  if (n > 0xFFFFFFFFLL) FAIL(GPUQ_ERR_INVALID, "partition: nrows %lld > 2^32", (long long)n);
  if (nparts < 1 || nparts > 65536)
    FAIL(GPUQ_ERR_INVALID, "partition: num_parts %d not in [1,65536]", nparts);
  if (key.dtype != GPUQ_INT64) FAIL(GPUQ_ERR_INVALID, "partition: key must be int64");
  sort_ws w; int64_t need;
  sort_ws_layout(n, 256, &w, (char*)workspace, &need);
  if (workspace_bytes < need)
    FAIL(GPUQ_ERR_INVALID, "partition: workspace %lld < %lld", (long long)workspace_bytes, (long long)need);
  HIP_TRY(hipMemsetAsync(out_counts, 0, (size_t)nparts * 8, s));
  if (n == 0) return GPUQ_OK;
  { hipEvent_t _pe = prof_begin(s);
    k_partition_pids<<<grid1d(n), 256, 0, s>>>(n, (const int64_t*)key.data, key.validity,
                                             nparts, w.ka, w.ia,
                                             (unsigned long long*)out_counts);
    prof_end("partition_pids", s, _pe); }
  HIP_TRY(hipGetLastError());
  scatter_geom geom = get_sort_geom();
  int tile = geom.block * geom.items;
  int64_t nb = sort_nblocks(n, tile);
  /* stable LSB radix over the pid: one 8-bit pass, two when nparts > 256 */
  int passes = nparts > 256 ? 2 : 1;
  uint64_t *kin = w.ka, *kout = w.kb;
  uint32_t *iin = w.ia, *iout = w.ib;
  for (int p = 0; p < passes; p++) {
    { hipEvent_t _pe = prof_begin(s);
    k_radix_hist<0><<<dim3((uint32_t)nb), 256, 0, s>>>(n, kin, p * 8, w.hist, (int)nb, tile);
    prof_end("radix_hist", s, _pe); }
    HIP_TRY(hipGetLastError());
    int rc = exclusive_scan_u32(s, (int64_t)256 * nb, w.hist, w.hist_scan, w.block_sums);
    if (rc) return rc;
    uint32_t* iout_pass = (p == passes - 1) ? out_perm : iout;
    { hipEvent_t _pe = prof_begin(s);
    launch_scatter<0, false>(s, geom, nb, n, kin, iin, kout, iout_pass, w.hist_scan,
                             p * 8, 0, nullptr, nullptr, nullptr, 0);
    prof_end("radix_scatter", s, _pe); }
    HIP_TRY(hipGetLastError());
    uint64_t* tk = kin; kin = kout; kout = tk;
    uint32_t* ti = iin; iin = iout_pass; iout = ti;
  }
  return GPUQ_OK;
}

/* ---- range partition: bin = first bound >= key (RangePartitioning,
 * exchange/ShuffleExchangeExec.scala:379-400: sampled bounds, keys <= 
 * bounds[j] stay in partition j; partition order follows the sort order,
 * so a per-rank sort after the exchange yields a globally ordered
 * rank-major result). Keys and bounds are radix-encoded with the SAME
 * encoding as the sort (desc = complement), so ascending logic covers
 * both directions. NULL keys go to the first (nulls_first) or last
 * partition, matching SortOrder null placement. nbounds <= 2047. */
__global__ void k_range_pids(int64_t n, const void* keys, const uint8_t* validity,
                             int dtype, int desc, int nulls_first,
                             const void* bounds, int nbounds,
                             uint64_t* pid_as_key, uint32_t* idx,
                             unsigned long long* counts /* [nbounds+1] */) {
  extern __shared__ __attribute__((aligned(16))) unsigned long long eb[];
  for (int j = threadIdx.x; j < nbounds; j += blockDim.x) {
    uint64_t e = (dtype == GPUQ_FLOAT64) ? encode_f64(((const double*)bounds)[j])
                                         : encode_i64(((const int64_t*)bounds)[j]);
    eb[j] = desc ? ~e : e;
  }
  __syncthreads();
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int pid;
    if (!bit_valid(validity, i)) {
      pid = nulls_first ? 0 : nbounds;
    } else {
      uint64_t e = (dtype == GPUQ_FLOAT64) ? encode_f64(((const double*)keys)[i])
                                           : encode_i64(((const int64_t*)keys)[i]);
      if (desc) e = ~e;
      int lo = 0, hi = nbounds;          /* first j with e <= eb[j] */
      while (lo < hi) {
        int mid = (lo + hi) >> 1;
        if (e <= eb[mid]) hi = mid; else lo = mid + 1;
      }
      pid = lo;
    }
    pid_as_key[i] = (uint64_t)pid;
    idx[i] = (uint32_t)i;
    atomicAdd(&counts[pid], 1ull);
  }
}

extern "C" int gpuq_range_partition_perm(void* stream, int64_t n, gpuq_col key,
                                         int32_t desc, int32_t nulls_first,
                                         const void* bounds, int32_t nbounds,
                                         uint32_t* out_perm, int64_t* out_counts,
                                         void* workspace, int64_t workspace_bytes) {
  hipStream_t s = (hipStream_t)stream;
  if (n > 0xFFFFFFFFLL) FAIL(GPUQ_ERR_INVALID, "range_partition: nrows %lld > 2^32", (long long)n);
  if (nbounds < 0 || nbounds > 2047)
    FAIL(GPUQ_ERR_INVALID, "range_partition: nbounds %d not in [0,2047]", nbounds);
  if (key.dtype != GPUQ_INT64 && key.dtype != GPUQ_FLOAT64)
    FAIL(GPUQ_ERR_INVALID, "range_partition: unsupported dtype %d", key.dtype);
  sort_ws w; int64_t need;
  sort_ws_layout(n, 256, &w, (char*)workspace, &need);
  if (workspace_bytes < need)
    FAIL(GPUQ_ERR_INVALID, "range_partition: workspace %lld < %lld",
         (long long)workspace_bytes, (long long)need);
  HIP_TRY(hipMemsetAsync(out_counts, 0, (size_t)(nbounds + 1) * 8, s));
  if (n == 0) return GPUQ_OK;
  k_range_pids<<<grid1d(n), 256, (uint32_t)(nbounds * 8), s>>>(
      n, key.data, key.validity, key.dtype, desc, nulls_first, bounds, nbounds,
      w.ka, w.ia, (unsigned long long*)out_counts);
  HIP_TRY(hipGetLastError());
  scatter_geom geom = get_sort_geom();
  int tile = geom.block * geom.items;
  int64_t nb = sort_nblocks(n, tile);
  int passes = (nbounds + 1) > 256 ? 2 : 1;
  uint64_t *kin = w.ka, *kout = w.kb;
  uint32_t *iin = w.ia, *iout = w.ib;
  for (int p = 0; p < passes; p++) {
    k_radix_hist<0><<<dim3((uint32_t)nb), 256, 0, s>>>(n, kin, p * 8, w.hist, (int)nb, tile);
    HIP_TRY(hipGetLastError());
    int rc = exclusive_scan_u32(s, (int64_t)256 * nb, w.hist, w.hist_scan, w.block_sums);
    if (rc) return rc;
    uint32_t* iout_pass = (p == passes - 1) ? out_perm : iout;
    launch_scatter<0, false>(s, geom, nb, n, kin, iin, kout, iout_pass, w.hist_scan,
                             p * 8, 0, nullptr, nullptr, nullptr, 0);
    HIP_TRY(hipGetLastError());
    uint64_t* tk = kin; kin = kout; kout = tk;
    uint32_t* ti = iin; iin = iout_pass; iout = ti;
  }
  return GPUQ_OK;
}

/* ================= hash aggregate ================= */
/*
 * Open-address table, EMPTY key sentinel = -1 (memset 0xFF); rows whose key
 * IS -1 and NULL-key rows use dedicated special slots so the full int64
 * domain is exact. Linear probing on Murmur3(key,42) (the reference probes
 * the same hash with triangular steps, BytesToBytesMap.java:513-539 —
 * probe order is unobservable in results).
 * Workspace: [cap u64 keys][cap f64 sums][cap u64 counts][special block]
 */

#define AGG_EMPTY 0xFFFFFFFFFFFFFFFFULL

struct agg_special {
  double m1_sum;   unsigned long long m1_cnt;  unsigned long long m1_seen;
  double nul_sum;  unsigned long long nul_cnt; unsigned long long nul_seen;
  unsigned long long out_cursor;
  unsigned long long overflow;
};

/* Interleaved slots [key, sum-bits, count(, pad)] so one probe touches one
 * cache line instead of three parallel arrays. Stride (u64 words/slot) is
 * 3 (24 B, default) or 4 (32 B, never line-straddling) via GPUQ_AGG_STRIDE. */
static int agg_stride(void) {
  static int s = 0;
  if (!s) {
    const char* e = getenv("GPUQ_AGG_STRIDE");
    s = (e && atoi(e) == 4) ? 4 : 3;
  }
  return s;
}

struct agg_ws {
  unsigned long long* tab;   /* stride * cap u64 */
  agg_special* sp;
};

static void agg_ws_layout(int64_t cap, agg_ws* w, char* base, int64_t* total) {
  int64_t off = 0;
  auto take = [&](int64_t bytes) {
    char* p = base ? base + off : nullptr;
    off += (bytes + 255) & ~255LL;
    return p;
  };
  w->tab = (unsigned long long*)take(cap * 8 * 4);  /* sized for stride 4 */
  w->sp = (agg_special*)take(sizeof(agg_special));
  *total = off;
}

extern "C" int64_t gpuq_hash_agg_workspace_bytes(int64_t cap) {
  agg_ws w; int64_t total;
  agg_ws_layout(cap, &w, nullptr, &total);
  return total;
}

template <int SLOT>
__global__ void k_agg_init(int64_t cap, unsigned long long* tab) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < cap; i += stride) {
    tab[SLOT * i] = AGG_EMPTY;
    tab[SLOT * i + 1] = 0;
    tab[SLOT * i + 2] = 0;
  }
}

#define AGG_OP_SUM 1
#define AGG_OP_COUNT 2

template <int OPS, int SLOT>
__global__ void k_agg_build(int64_t n, const int64_t* keys, const uint8_t* kvalid,
                            const double* vals, const uint8_t* vvalid,
                            unsigned long long* tab, agg_special* sp,
                            int64_t cap_mask) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    bool kv = bit_valid(kvalid, i);
    bool vv = bit_valid(vvalid, i);
    double v = vv ? vals[i] : 0.0;
    if (!kv || (unsigned long long)keys[i] == AGG_EMPTY) {
      double* psum = kv ? &sp->m1_sum : &sp->nul_sum;
      unsigned long long* pcnt = kv ? &sp->m1_cnt : &sp->nul_cnt;
      unsigned long long* pseen = kv ? &sp->m1_seen : &sp->nul_seen;
      atomicMax(pseen, 1ull);
      if (vv) {
        if (OPS & AGG_OP_SUM) atomicAdd(psum, v);
        if (OPS & AGG_OP_COUNT) atomicAdd(pcnt, 1ull);
      }
      continue;
    }
    int64_t k = keys[i];
    uint64_t slot = ((uint32_t)mm3_hash_long(k, 42)) & (uint64_t)cap_mask;
    for (int probes = 0;; probes++) {
      unsigned long long cur = __hip_atomic_load(&tab[SLOT * slot], __ATOMIC_RELAXED,
                                                 __HIP_MEMORY_SCOPE_AGENT);
      if (cur == (unsigned long long)k) break;
      if (cur == AGG_EMPTY) {
        unsigned long long prev = atomicCAS(&tab[SLOT * slot], AGG_EMPTY, (unsigned long long)k);
        if (prev == AGG_EMPTY || prev == (unsigned long long)k) break;
      }
      slot = (slot + 1) & (uint64_t)cap_mask;
      if (probes > cap_mask) { atomicMax(&sp->overflow, 1ull); return; }
    }
    if (vv) {
      if (OPS & AGG_OP_SUM) atomicAdd((double*)&tab[SLOT * slot + 1], v);
      if (OPS & AGG_OP_COUNT) atomicAdd(&tab[SLOT * slot + 2], 1ull);
    }
  }
}

#define AGGC_CHUNK 8192

template <int OPS, int SLOT>
__global__ void k_agg_compact(int64_t cap, const unsigned long long* tab,
                              agg_special* sp,
                              int64_t* out_keys, uint8_t* out_kvalid,
                              double* out_sums, uint8_t* out_svalid,
                              int64_t* out_cnts) {
  /* ONE output-cursor atomic per block (a single cursor word saturates
   * near ~88 updates/us): count occupied slots in this block's chunk,
   * block-scan, reserve once, emit. */
  constexpr int ROUNDS = AGGC_CHUNK / 256;
  __shared__ unsigned long long block_base;
  __shared__ uint32_t wtot[4];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int64_t base = (int64_t)blockIdx.x * AGGC_CHUNK;
  uint32_t occ_mask[ROUNDS / 32 + 1];
  for (int m = 0; m < ROUNDS / 32 + 1; m++) occ_mask[m] = 0;
  uint32_t lane_total = 0;
  for (int r = 0; r < ROUNDS; r++) {
    int64_t i = base + r * 256 + threadIdx.x;
    bool occ = i < cap && tab[SLOT * i] != AGG_EMPTY;
    if (occ) { occ_mask[r / 32] |= 1u << (r & 31); lane_total++; }
  }
  uint32_t incl = wave_inclusive_scan(lane_total);
  if (lane == WAVE - 1) wtot[wave] = incl;
  __syncthreads();
  if (threadIdx.x == 0) {
    uint32_t tot = 0;
    for (int w = 0; w < 4; w++) { uint32_t t = wtot[w]; wtot[w] = tot; tot += t; }
    block_base = tot ? atomicAdd(&sp->out_cursor, (unsigned long long)tot) : 0;
  }
  __syncthreads();
  int64_t o = (int64_t)block_base + wtot[wave] + (incl - lane_total);
  for (int r = 0; r < ROUNDS; r++) {
    if (!((occ_mask[r / 32] >> (r & 31)) & 1)) continue;
    int64_t i = base + r * 256 + threadIdx.x;
    unsigned long long k = tab[SLOT * i];
    out_keys[o] = (int64_t)k;
    out_kvalid[o] = 1;
    out_sums[o] = __longlong_as_double((long long)tab[SLOT * i + 1]);
    out_svalid[o] = (OPS & AGG_OP_COUNT) ? (tab[SLOT * i + 2] > 0 ? 1 : 0) : 1;
    if (out_cnts) out_cnts[o] = (int64_t)tab[SLOT * i + 2];
    o++;
  }
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    if (sp->m1_seen) {
      int64_t q = (int64_t)atomicAdd(&sp->out_cursor, 1ull);
      out_keys[q] = -1; out_kvalid[q] = 1;
      out_sums[q] = sp->m1_sum;
      out_svalid[q] = (OPS & AGG_OP_COUNT) ? (sp->m1_cnt > 0 ? 1 : 0) : 1;
      if (out_cnts) out_cnts[q] = (int64_t)sp->m1_cnt;
    }
    if (sp->nul_seen) {
      int64_t q = (int64_t)atomicAdd(&sp->out_cursor, 1ull);
      out_keys[q] = 0; out_kvalid[q] = 0;
      out_sums[q] = sp->nul_sum;
      out_svalid[q] = (OPS & AGG_OP_COUNT) ? (sp->nul_cnt > 0 ? 1 : 0) : 1;
      if (out_cnts) out_cnts[q] = (int64_t)sp->nul_cnt;
    }
  }
}

/* small-cardinality build: per-block LDS table (classic two-level
 * aggregation), merged once into the global table per block. Removes the
 * global-atomic contention wall on few-group aggregates (measured 607 ms
 * for 1B rows / 1K groups on the direct path). cap <= AGG_LDS_CAP slots. */
#define AGG_LDS_CAP 2048

template <int OPS, int SLOT>
__global__ __launch_bounds__(256)
void k_agg_build_lds(int64_t n, const int64_t* keys, const uint8_t* kvalid,
                     const double* vals, const uint8_t* vvalid,
                     unsigned long long* tab, agg_special* sp,
                     int64_t cap_mask) {
  __shared__ unsigned long long lt[AGG_LDS_CAP * 3];
  const int64_t lcap = cap_mask + 1;
  for (int j = threadIdx.x; j < (int)lcap; j += blockDim.x) {
    lt[3 * j] = AGG_EMPTY; lt[3 * j + 1] = 0; lt[3 * j + 2] = 0;
  }
  __syncthreads();
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    bool kv = bit_valid(kvalid, i);
    bool vv = bit_valid(vvalid, i);
    double v = vv ? vals[i] : 0.0;
    if (!kv || (unsigned long long)keys[i] == AGG_EMPTY) {
      double* psum = kv ? &sp->m1_sum : &sp->nul_sum;
      unsigned long long* pcnt = kv ? &sp->m1_cnt : &sp->nul_cnt;
      unsigned long long* pseen = kv ? &sp->m1_seen : &sp->nul_seen;
      atomicMax(pseen, 1ull);
      if (vv) {
        if (OPS & AGG_OP_SUM) atomicAdd(psum, v);
        if (OPS & AGG_OP_COUNT) atomicAdd(pcnt, 1ull);
      }
      continue;
    }
    int64_t k = keys[i];
    uint64_t slot = ((uint32_t)mm3_hash_long(k, 42)) & (uint64_t)cap_mask;
    for (int probes = 0;; probes++) {
      unsigned long long cur = lt[3 * slot];
      if (cur == (unsigned long long)k) break;
      if (cur == AGG_EMPTY) {
        unsigned long long prev = atomicCAS(&lt[3 * slot], AGG_EMPTY,
                                            (unsigned long long)k);
        if (prev == AGG_EMPTY || prev == (unsigned long long)k) break;
      }
      slot = (slot + 1) & (uint64_t)cap_mask;
      if (probes > cap_mask) { atomicMax(&sp->overflow, 1ull); return; }
    }
    if (vv) {
      if (OPS & AGG_OP_SUM) atomicAdd((double*)&lt[3 * slot + 1], v);
      if (OPS & AGG_OP_COUNT) atomicAdd(&lt[3 * slot + 2], 1ull);
    }
  }
  __syncthreads();
  /* merge this block's LDS table into the global one (same probe scheme) */
  for (int j = threadIdx.x; j < (int)lcap; j += blockDim.x) {
    unsigned long long k = lt[3 * j];
    if (k == AGG_EMPTY) continue;
    uint64_t slot = ((uint32_t)mm3_hash_long((int64_t)k, 42)) & (uint64_t)cap_mask;
    for (;;) {
      unsigned long long cur = __hip_atomic_load(&tab[SLOT * slot], __ATOMIC_RELAXED,
                                                 __HIP_MEMORY_SCOPE_AGENT);
      if (cur == k) break;
      if (cur == AGG_EMPTY) {
        unsigned long long prev = atomicCAS(&tab[SLOT * slot], AGG_EMPTY, k);
        if (prev == AGG_EMPTY || prev == k) break;
      }
      slot = (slot + 1) & (uint64_t)cap_mask;
    }
    if (OPS & AGG_OP_SUM)
      atomicAdd((double*)&tab[SLOT * slot + 1],
                __longlong_as_double((long long)lt[3 * j + 1]));
    if (OPS & AGG_OP_COUNT) atomicAdd(&tab[SLOT * slot + 2], lt[3 * j + 2]);
  }
}

extern "C" int gpuq_hash_agg_i64_f64(void* stream, int64_t n,
                                     gpuq_col key, gpuq_col val,
                                     void* workspace, int64_t cap, int32_t first_batch,
                                     int32_t finalize, int32_t ops,
                                     int64_t* out_keys, uint8_t* out_key_valid,
                                     double* out_sums, uint8_t* out_sum_valid,
                                     int64_t* out_counts, int64_t* out_ngroups) {
  hipStream_t s = (hipStream_t)stream;
  if (cap <= 0 || (cap & (cap - 1)))
    FAIL(GPUQ_ERR_INVALID, "agg: capacity %lld not a power of two", (long long)cap);
  if (key.dtype != GPUQ_INT64 || val.dtype != GPUQ_FLOAT64)
    FAIL(GPUQ_ERR_INVALID, "agg: expected int64 key + float64 val");
  if (!(ops & AGG_OP_COUNT) && val.validity)
    FAIL(GPUQ_ERR_INVALID, "agg: SUM-only mode requires non-null values "
         "(sum NULL-ness needs COUNT)");
  agg_ws w; int64_t need;
  agg_ws_layout(cap, &w, (char*)workspace, &need);
  if (first_batch) {
    if (agg_stride() == 4) k_agg_init<4><<<grid1d(cap), 256, 0, s>>>(cap, w.tab);
    else k_agg_init<3><<<grid1d(cap), 256, 0, s>>>(cap, w.tab);
    HIP_TRY(hipGetLastError());
    HIP_TRY(hipMemsetAsync(w.sp, 0, sizeof(agg_special), s));
  }
  if (n > 0) {
    { hipEvent_t _pe = prof_begin(s);
    bool lds_path = cap <= AGG_LDS_CAP && getenv("GPUQ_NO_LDS_AGG") == nullptr;
#define AGB(OPS, SL) k_agg_build<OPS, SL><<<hash_grid(n), 256, 0, s>>>( \
        n, (const int64_t*)key.data, key.validity, \
        (const double*)val.data, val.validity, w.tab, w.sp, cap - 1)
#define AGBL(OPS, SL) k_agg_build_lds<OPS, SL><<<hash_grid(n), 256, 0, s>>>( \
        n, (const int64_t*)key.data, key.validity, \
        (const double*)val.data, val.validity, w.tab, w.sp, cap - 1)
    if (lds_path) {
      if (agg_stride() == 4) { if (ops == AGG_OP_SUM) AGBL(AGG_OP_SUM, 4); else AGBL(3, 4); }
      else { if (ops == AGG_OP_SUM) AGBL(AGG_OP_SUM, 3); else AGBL(3, 3); }
    } else {
      if (agg_stride() == 4) { if (ops == AGG_OP_SUM) AGB(AGG_OP_SUM, 4); else AGB(3, 4); }
      else { if (ops == AGG_OP_SUM) AGB(AGG_OP_SUM, 3); else AGB(3, 3); }
    }
#undef AGB
#undef AGBL
    prof_end("agg_build", s, _pe); }
    HIP_TRY(hipGetLastError());
  }
  if (finalize) {
    agg_special hsp;
    HIP_TRY(hipMemcpyAsync(&hsp, w.sp, sizeof(hsp), hipMemcpyDeviceToHost, s));
    HIP_TRY(hipStreamSynchronize(s));
    if (hsp.overflow) FAIL(GPUQ_ERR_OVERFLOW, "agg: hash table overflow (capacity %lld)", (long long)cap);
    { hipEvent_t _pe = prof_begin(s);
    dim3 cgrid((uint32_t)((cap + AGGC_CHUNK - 1) / AGGC_CHUNK));
#define AGC(OPS, SL) k_agg_compact<OPS, SL><<<cgrid, 256, 0, s>>>( \
        cap, w.tab, w.sp, out_keys, out_key_valid, out_sums, out_sum_valid, \
        out_counts)
    if (agg_stride() == 4) { if (ops == AGG_OP_SUM) AGC(AGG_OP_SUM, 4); else AGC(3, 4); }
    else { if (ops == AGG_OP_SUM) AGC(AGG_OP_SUM, 3); else AGC(3, 3); }
#undef AGC
    prof_end("agg_compact", s, _pe); }
    HIP_TRY(hipGetLastError());
    agg_special hsp2;
    HIP_TRY(hipMemcpyAsync(&hsp2, w.sp, sizeof(hsp2), hipMemcpyDeviceToHost, s));
    HIP_TRY(hipStreamSynchronize(s));
    *out_ngroups = (int64_t)hsp2.out_cursor;
  }
  return GPUQ_OK;
}

/* ================= hash join ================= */
/*
 * Build: table of 16-byte interleaved slots {key u64, head u32, pad};
 * duplicates chained through next[] (the GPU analog of LongToUnsafeRowMap's
 * per-row next pointer, HashedRelation.scala:536-600), chain heads carry a
 * MULTI bit so single-match probes never read next[]. EMPTY key sentinel -1
 * with a dedicated chain for real -1 keys. NULL keys never match.
 *
 * Locality design (MI355X: random HBM lines are ~7x slower than L3-resident
 * access — measured in tools/diag_hash.py): the slot index is the TOP bits
 * of Murmur3(key,42) (multiplicative map, monotone in hash), and when keys
 * have no NULLs both sides are first radix-partitioned into hash order
 * (one ranked-scatter pass, BIN_MODE 2). Blocks then process contiguous
 * chunks of the hash-ordered stream, so the live table region at any
 * instant is a small hash-contiguous window that stays L3-resident, and
 * next[]/rid side arrays are bucket-local. Probe reserves output space
 * with one wave-aggregated atomic.
 */

#define JOIN_NIL 0xFFFFFFFFu
#ifndef JOIN_CHUNK
#define JOIN_CHUNK 4096
#endif

struct join_sp { unsigned int m1_head; unsigned int bucketed; unsigned long long cursor; };

/* hash -> slot: monotone multiplicative map onto [0, cap) */
DEV uint64_t join_slot(int64_t key, int64_t cap_mask) {
  uint32_t h = (uint32_t)mm3_hash_long(key, 42);
  return ((uint64_t)h * (uint64_t)(cap_mask + 1)) >> 32;
}

struct join_ws {
  unsigned long long* slots;   /* 2 u64 per slot: [key][head|pad] */
  unsigned int* next;
  unsigned long long* matched; /* build-row matched bits (full outer) */
  unsigned int* brid;          /* partitioned row id map (bucketed path) */
  uint64_t* pk_a; uint32_t* pi_a;   /* partition scratch: pairs in */
  uint64_t* pk_b;                   /* partitioned keys out (ids go to brid) */
  uint32_t* hist; uint32_t* hist_scan; uint32_t* block_sums;
  join_sp* sp;
};

static void join_ws_layout(int64_t cap, int64_t brows, join_ws* w, char* base, int64_t* total) {
  int tile = get_sort_geom().block * get_sort_geom().items;
  int64_t nb = sort_nblocks(brows, tile);
  int64_t hist_n = 256 * nb;
  int64_t scan_blocks = (hist_n + SCAN_TILE - 1) / SCAN_TILE + 1;
  int64_t off = 0;
  auto take = [&](int64_t bytes) {
    char* p = base ? base + off : nullptr;
    off += (bytes + 255) & ~255LL;
    return p;
  };
  w->slots = (unsigned long long*)take(cap * 16);
  w->next = (unsigned int*)take(brows * 4);
  w->matched = (unsigned long long*)take(((brows + 63) / 64) * 8);
  w->brid = (unsigned int*)take(brows * 4);
  w->pk_a = (uint64_t*)take(brows * 8);
  w->pi_a = (uint32_t*)take(brows * 4);
  w->pk_b = (uint64_t*)take(brows * 8);
  w->hist = (uint32_t*)take(hist_n * 4);
  w->hist_scan = (uint32_t*)take(hist_n * 4);
  w->block_sums = (uint32_t*)take(scan_blocks * 4);
  w->sp = (join_sp*)take(sizeof(join_sp));
  *total = off;
}

extern "C" int64_t gpuq_join_build_workspace_bytes(int64_t brows, int64_t cap) {
  join_ws w; int64_t total;
  join_ws_layout(cap, brows, &w, nullptr, &total);
  return total;
}

/* probe-side scratch (hash-ordered probe stream) */
struct probe_ws {
  uint64_t* pk_a; uint32_t* pi_a;
  uint64_t* pk_b; uint32_t* pi_b;
  uint32_t* hist; uint32_t* hist_scan; uint32_t* block_sums;
};

static void probe_ws_layout(int64_t prows, probe_ws* w, char* base, int64_t* total) {
  int tile = get_sort_geom().block * get_sort_geom().items;
  int64_t nb = sort_nblocks(prows, tile);
  int64_t hist_n = 256 * nb;
  int64_t scan_blocks = (hist_n + SCAN_TILE - 1) / SCAN_TILE + 1;
  int64_t off = 0;
  auto take = [&](int64_t bytes) {
    char* p = base ? base + off : nullptr;
    off += (bytes + 255) & ~255LL;
    return p;
  };
  w->pk_a = (uint64_t*)take(prows * 8);
  w->pi_a = (uint32_t*)take(prows * 4);
  w->pk_b = (uint64_t*)take(prows * 8);
  w->pi_b = (uint32_t*)take(prows * 4);
  w->hist = (uint32_t*)take(hist_n * 4);
  w->hist_scan = (uint32_t*)take(hist_n * 4);
  w->block_sums = (uint32_t*)take(scan_blocks * 4);
  *total = off;
}

extern "C" int64_t gpuq_join_probe_workspace_bytes(int64_t prows) {
  probe_ws w; int64_t total;
  probe_ws_layout(prows, &w, nullptr, &total);
  return total;
}

__global__ void k_make_pairs(int64_t n, const int64_t* keys, uint64_t* out_k,
                             uint32_t* out_i) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    out_k[i] = (uint64_t)keys[i];
    out_i[i] = (uint32_t)i;
  }
}

/* one hash-order partition pass over (key, rowid) pairs */
static int hash_order_pairs(hipStream_t s, int64_t n, const int64_t* keys,
                            uint64_t* tmp_k, uint32_t* tmp_i,
                            uint64_t* out_k, uint32_t* out_i,
                            uint32_t* hist, uint32_t* hist_scan,
                            uint32_t* block_sums) {
  scatter_geom geom = get_sort_geom();
  int tile = geom.block * geom.items;
  int64_t nb = sort_nblocks(n, tile);
  k_make_pairs<<<grid1d(n), 256, 0, s>>>(n, keys, tmp_k, tmp_i);
  HIP_TRY(hipGetLastError());
  { hipEvent_t _pe = prof_begin(s);
  k_radix_hist<2><<<dim3((uint32_t)nb), 256, 0, s>>>(n, tmp_k, 0, hist, (int)nb, tile);
  prof_end("join_part_hist", s, _pe); }
  HIP_TRY(hipGetLastError());
  int rc = exclusive_scan_u32(s, 256 * nb, hist, hist_scan, block_sums);
  if (rc) return rc;
  { hipEvent_t _pe = prof_begin(s);
  launch_scatter<2, false>(s, geom, nb, n, tmp_k, tmp_i, out_k, out_i,
                           hist_scan, 0, 0, nullptr, nullptr, nullptr, 0);
  prof_end("join_part_scatter", s, _pe); }
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

/* chain-head word: bit 31 = MULTI (slot holds >1 row) so single-match
 * probes never touch next[]; low 31 bits = first build row index */
#define JOIN_MULTI 0x80000000u

DEV void join_push_head(unsigned int* headp, unsigned int i, unsigned int* next) {
  unsigned int old = __hip_atomic_load(headp, __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT);
  for (;;) {
    unsigned int newv = i | (old != JOIN_NIL ? JOIN_MULTI : 0u);
    unsigned int prev = atomicCAS(headp, old, newv);
    if (prev == old) {
      next[i] = (old == JOIN_NIL) ? JOIN_NIL : (old & ~JOIN_MULTI);
      break;
    }
    old = prev;
  }
}

/* contiguous-chunk mapping: block b owns rows [b*CHUNK, (b+1)*CHUNK) so the
 * live window of a hash-ordered stream is hash-contiguous (L3-resident) */
__global__ void k_join_build(int64_t n, const int64_t* keys, const uint8_t* kvalid,
                             unsigned long long* slots, unsigned int* next,
                             join_sp* sp, int64_t cap_mask) {
  int64_t base = (int64_t)blockIdx.x * JOIN_CHUNK;
  for (int r = 0; r < JOIN_CHUNK / 256; r++) {
    int64_t i = base + r * 256 + threadIdx.x;
    if (i >= n) return;
    if (!bit_valid(kvalid, i)) continue;  /* NULL never matches (inner join) */
    int64_t k = keys[i];
    if ((unsigned long long)k == AGG_EMPTY) {
      join_push_head(&sp->m1_head, (unsigned int)i, next);
      continue;
    }
    uint64_t slot = join_slot(k, cap_mask);
    for (;;) {
      unsigned long long cur = __hip_atomic_load(&slots[2 * slot], __ATOMIC_RELAXED,
                                                 __HIP_MEMORY_SCOPE_AGENT);
      if (cur == (unsigned long long)k) break;
      if (cur == AGG_EMPTY) {
        unsigned long long prev = atomicCAS(&slots[2 * slot], AGG_EMPTY, (unsigned long long)k);
        if (prev == AGG_EMPTY || prev == (unsigned long long)k) break;
      }
      slot = (slot + 1) & (uint64_t)cap_mask;
    }
    join_push_head((unsigned int*)&slots[2 * slot + 1], (unsigned int)i, next);
  }
}

/* join types over the PROBE (streamed) side — the reference's
 * ShuffledHashJoinExec joinType dispatch (ShuffledHashJoinExec.scala /
 * HashJoin.scala join() : inner, outer, semi, anti):
 * 0 = Inner, 1 = probe-side Outer (unmatched probe rows emit one pair
 * with build rid = JOIN_NIL -> NULL build columns), 2 = LeftSemi (probe
 * row emitted once iff matched), 3 = LeftAnti (emitted iff unmatched;
 * NULL probe keys never match, so they emit — the non-null-aware anti). */
DEV uint32_t jt_emit_count(int jt, uint32_t m) {
  if (jt == 1 || jt == 4) return m ? m : 1u;  /* 4 = FullOuter probe half */
  if (jt == 2) return m ? 1u : 0u;
  if (jt == 3 || jt == 5) return m ? 0u : 1u; /* 5 = null-aware anti */
  return m;
}

template <int JT>
__global__ void k_join_probe(int64_t n, const int64_t* keys, const uint8_t* kvalid,
                             const unsigned long long* slots, const unsigned int* next,
                             const unsigned int* brid_map, const unsigned int* prid_map,
                             join_sp* sp, int64_t cap_mask,
                             uint32_t* out_p, uint32_t* out_b, int64_t out_cap,
                             unsigned long long* matched = nullptr) {
  constexpr int ROUNDS = JOIN_CHUNK / 256;
  __shared__ unsigned long long block_base;
  __shared__ uint32_t wtot[JOIN_CHUNK / 256 > 4 ? 16 : 16];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int64_t base = (int64_t)blockIdx.x * JOIN_CHUNK;

  /* phase A: all lookups; match counts + first two matches stay in
   * registers. ONE output-cursor atomic per block (a single global cursor
   * word saturates near ~88 updates/us — the per-wave-iteration version
   * was the whole kernel's bottleneck). */
  uint32_t cnt[ROUNDS];
  unsigned int c0[ROUNDS], c1[ROUNDS];
  uint32_t lane_total = 0;
  for (int r = 0; r < ROUNDS; r++) {
    int64_t i = base + r * 256 + threadIdx.x;
    unsigned int head = JOIN_NIL;
    if (i < n && bit_valid(kvalid, i)) {
      int64_t k = keys[i];
      if ((unsigned long long)k == AGG_EMPTY) {
        head = sp->m1_head;
      } else {
        uint64_t slot = join_slot(k, cap_mask);
        for (;;) {
          unsigned long long cur = slots[2 * slot];
          if (cur == AGG_EMPTY) break;
          if (cur == (unsigned long long)k) {
            head = (unsigned int)slots[2 * slot + 1];
            break;
          }
          slot = (slot + 1) & (uint64_t)cap_mask;
        }
      }
    }
    cnt[r] = 0; c0[r] = JOIN_NIL; c1[r] = JOIN_NIL;
    if (head != JOIN_NIL) {
      if (!(head & JOIN_MULTI)) {
        c0[r] = head; cnt[r] = 1;
        if (JT == 4) atomicOr(&matched[head >> 6], 1ull << (head & 63));
      } else {
        for (unsigned int b = head & ~JOIN_MULTI; b != JOIN_NIL; b = next[b]) {
          if (cnt[r] == 0) c0[r] = b; else if (cnt[r] == 1) c1[r] = b;
          cnt[r]++;
          if (JT == 4) atomicOr(&matched[b >> 6], 1ull << (b & 63));
        }
      }
    }
    if (JT == 5 && i < n && !bit_valid(kvalid, i))
      cnt[r] = 1;  /* null-aware anti: NULL NOT IN (non-empty) is unknown
                    * -> the row is filtered (treated as matched);
                    * BroadcastHashJoinExec.scala:137 NAAJ semantics */
    if (JT == 0) lane_total += cnt[r];
    else if (i < n) lane_total += jt_emit_count(JT, cnt[r]);
  }
  /* phase B: block-level reservation (emit order within the block is
   * arbitrary — join output order is nondeterministic by contract) */
  uint32_t incl = wave_inclusive_scan(lane_total);
  if (lane == WAVE - 1) wtot[wave] = incl;
  __syncthreads();
  if (threadIdx.x == 0) {
    uint32_t tot = 0;
    for (int w = 0; w < (int)(blockDim.x / WAVE); w++) {
      uint32_t t = wtot[w]; wtot[w] = tot; tot += t;
    }
    block_base = tot ? atomicAdd(&sp->cursor, (unsigned long long)tot) : 0;
  }
  __syncthreads();
  int64_t o = (int64_t)block_base + wtot[wave] + (incl - lane_total);
  for (int r = 0; r < ROUNDS; r++) {
    int64_t i = base + r * 256 + threadIdx.x;
    if (JT != 0) {
      if (i >= n || !jt_emit_count(JT, cnt[r])) continue;
      uint32_t pr = prid_map ? prid_map[i] : (uint32_t)i;
      if (JT == 2 || JT == 3 || JT == 5 || cnt[r] == 0) {
        /* semi/anti emit the probe row once; outer's unmatched row pairs
         * with NIL (NULL build columns downstream) */
        if (o < out_cap) { out_p[o] = pr; out_b[o] = JOIN_NIL; }
        o++;
        continue;
      }
    }
    if (!cnt[r]) continue;
    uint32_t pr = prid_map ? prid_map[i] : (uint32_t)i;
    if (o < out_cap) { out_p[o] = pr; out_b[o] = brid_map ? brid_map[c0[r]] : c0[r]; }
    o++;
    if (cnt[r] >= 2) {
      if (o < out_cap) { out_p[o] = pr; out_b[o] = brid_map ? brid_map[c1[r]] : c1[r]; }
      o++;
      unsigned int b = (cnt[r] > 2) ? next[c1[r]] : JOIN_NIL;
      for (uint32_t j = 2; j < cnt[r]; j++, b = next[b], o++) {
        if (o < out_cap) { out_p[o] = pr; out_b[o] = brid_map ? brid_map[b] : b; }
      }
    }
  }
}

/* FullOuter second half: one (NIL, build_rid) pair per unmatched build
 * row — every build row never touched by a probe (incl. NULL-key build
 * rows, which are never inserted and so never matched). */
__global__ void k_join_unmatched_build(int64_t brows,
                                       const unsigned long long* matched,
                                       const unsigned int* brid_map,
                                       join_sp* sp,
                                       uint32_t* out_p, uint32_t* out_b,
                                       int64_t out_cap) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  const int lane = threadIdx.x & (WAVE - 1);
  for (; i - lane < brows; i += gs) {
    bool un = i < brows && !((matched[i >> 6] >> (i & 63)) & 1);
    uint64_t m = __ballot(un);
    if (!m) continue;
    int cntw = __popcll(m);
    unsigned long long base = 0;
    int leader = __ffsll((unsigned long long)m) - 1;
    if (lane == leader)
      base = atomicAdd(&sp->cursor, (unsigned long long)cntw);
    base = __shfl(base, leader);
    if (!un) continue;
    int64_t o = (int64_t)base + __popcll(m & ((1ULL << lane) - 1));
    if (o < out_cap) {
      out_p[o] = JOIN_NIL;
      out_b[o] = brid_map ? brid_map[i] : (unsigned int)i;
    }
  }
}

extern "C" int gpuq_join_build_i64(void* stream, int64_t brows, gpuq_col bkey,
                                   void* workspace, int64_t cap) {
  hipStream_t s = (hipStream_t)stream;
  if (cap <= 0 || (cap & (cap - 1)))
    FAIL(GPUQ_ERR_INVALID, "join: capacity %lld not a power of two", (long long)cap);
  if (brows > 0x7FFFFFFELL) FAIL(GPUQ_ERR_INVALID, "join: build side too large for 31-bit rowids");
  if (bkey.dtype != GPUQ_INT64) FAIL(GPUQ_ERR_INVALID, "join: key must be int64");
  join_ws w; int64_t need;
  join_ws_layout(cap, brows, &w, (char*)workspace, &need);
  HIP_TRY(hipMemsetAsync(w.slots, 0xFF, cap * 16, s));  /* keys=-1, heads=NIL */
  HIP_TRY(hipMemsetAsync(w.matched, 0, ((brows + 63) / 64) * 8, s));
  HIP_TRY(hipMemsetAsync(w.sp, 0xFF, 4, s));            /* m1_head = NIL */
  HIP_TRY(hipMemsetAsync(&w.sp->bucketed, 0, 4, s));
  HIP_TRY(hipMemsetAsync(&w.sp->cursor, 0, 8, s));
  if (brows == 0) return GPUQ_OK;
  /* bucketed (hash-ordered) path when there are no NULLs and the table is
   * past L3 size; flat otherwise */
  /* hash-order bucketing measured as net overhead once the output-cursor
   * bottleneck was removed (probe is latency-, not line-refetch-bound);
   * kept behind an opt-in env for future asymmetric shapes */
  bool bucketed = bkey.validity == nullptr && cap * 16 > (256LL << 20) &&
                  getenv("GPUQ_BUCKET_JOIN") != nullptr;
  const int64_t* keys = (const int64_t*)bkey.data;
  if (bucketed) {
    int rc = hash_order_pairs(s, brows, keys, w.pk_a, w.pi_a, w.pk_b, w.brid,
                              w.hist, w.hist_scan, w.block_sums);
    if (rc) return rc;
    keys = (const int64_t*)w.pk_b;
    HIP_TRY(hipMemsetAsync(&w.sp->bucketed, 1, 1, s));
  }
  int64_t nchunks = (brows + JOIN_CHUNK - 1) / JOIN_CHUNK;
  { hipEvent_t _pe = prof_begin(s);
  k_join_build<<<dim3((uint32_t)nchunks), 256, 0, s>>>(
      brows, keys, bucketed ? nullptr : bkey.validity, w.slots, w.next, w.sp,
      cap - 1);
  prof_end("join_build", s, _pe); }
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

extern "C" int gpuq_join_probe_i64_typed(void* stream, int64_t prows, gpuq_col pkey,
                                   const void* workspace, int64_t cap, int64_t brows,
                                   void* probe_workspace, int64_t probe_ws_bytes,
                                   int32_t join_type,
                                   uint32_t* out_p, uint32_t* out_b,
                                   int64_t out_cap, int64_t* out_nmatches) {
  hipStream_t s = (hipStream_t)stream;
  if (join_type < 0 || join_type > 5)
    FAIL(GPUQ_ERR_INVALID, "join: bad join_type %d", join_type);
  if (pkey.dtype != GPUQ_INT64) FAIL(GPUQ_ERR_INVALID, "join: key must be int64");
  join_ws w; int64_t need;
  join_ws_layout(cap, brows, &w, (char*)workspace, &need);
  HIP_TRY(hipMemsetAsync(&w.sp->cursor, 0, 8, s));
  if (prows > 0) {
    join_sp hsp0;
    HIP_TRY(hipMemcpyAsync(&hsp0, w.sp, sizeof(hsp0), hipMemcpyDeviceToHost, s));
    HIP_TRY(hipStreamSynchronize(s));
    bool build_bucketed = hsp0.bucketed != 0;
    const int64_t* pkeys = (const int64_t*)pkey.data;
    const uint8_t* pvalid = pkey.validity;
    const unsigned int* prid_map = nullptr;
    probe_ws pw;
    /* order the probe stream by hash too (same locality argument) */
    bool probe_bucketed = build_bucketed && pvalid == nullptr &&
                          probe_workspace != nullptr;
    if (probe_bucketed) {
      int64_t pneed;
      probe_ws_layout(prows, &pw, (char*)probe_workspace, &pneed);
      if (probe_ws_bytes < pneed)
        FAIL(GPUQ_ERR_INVALID, "join: probe workspace %lld < %lld",
             (long long)probe_ws_bytes, (long long)pneed);
      int rc = hash_order_pairs(s, prows, pkeys, pw.pk_a, pw.pi_a, pw.pk_b,
                                pw.pi_b, pw.hist, pw.hist_scan, pw.block_sums);
      if (rc) return rc;
      pkeys = (const int64_t*)pw.pk_b;
      prid_map = pw.pi_b;
      pvalid = nullptr;
    }
    int64_t nchunks = (prows + JOIN_CHUNK - 1) / JOIN_CHUNK;
    { hipEvent_t _pe = prof_begin(s);
    const unsigned int* bm = build_bucketed ? w.brid : nullptr;
    dim3 jg((uint32_t)nchunks);
    if (join_type == 1)
      k_join_probe<1><<<jg, 256, 0, s>>>(prows, pkeys, pvalid, w.slots, w.next,
                                         bm, prid_map, w.sp, cap - 1, out_p, out_b, out_cap);
    else if (join_type == 2)
      k_join_probe<2><<<jg, 256, 0, s>>>(prows, pkeys, pvalid, w.slots, w.next,
                                         bm, prid_map, w.sp, cap - 1, out_p, out_b, out_cap);
    else if (join_type == 3)
      k_join_probe<3><<<jg, 256, 0, s>>>(prows, pkeys, pvalid, w.slots, w.next,
                                         bm, prid_map, w.sp, cap - 1, out_p, out_b, out_cap);
    else if (join_type == 4)
      k_join_probe<4><<<jg, 256, 0, s>>>(prows, pkeys, pvalid, w.slots, w.next,
                                         bm, prid_map, w.sp, cap - 1, out_p, out_b, out_cap,
                                         w.matched);
    else if (join_type == 5)
      k_join_probe<5><<<jg, 256, 0, s>>>(prows, pkeys, pvalid, w.slots, w.next,
                                         bm, prid_map, w.sp, cap - 1, out_p, out_b, out_cap);
    else
      k_join_probe<0><<<jg, 256, 0, s>>>(prows, pkeys, pvalid, w.slots, w.next,
                                         bm, prid_map, w.sp, cap - 1, out_p, out_b, out_cap);
    prof_end("join_probe", s, _pe); }
    HIP_TRY(hipGetLastError());
  }
  if (join_type == 4 && brows > 0) {
    /* FullOuter second half; probes with zero rows still emit every
     * build row */
    join_sp hsp0b;
    HIP_TRY(hipMemcpyAsync(&hsp0b, w.sp, sizeof(hsp0b), hipMemcpyDeviceToHost, s));
    HIP_TRY(hipStreamSynchronize(s));
    k_join_unmatched_build<<<grid1d(brows), 256, 0, s>>>(
        brows, w.matched, hsp0b.bucketed ? w.brid : nullptr, w.sp,
        out_p, out_b, out_cap);
    HIP_TRY(hipGetLastError());
  }
  join_sp hsp;
  HIP_TRY(hipMemcpyAsync(&hsp, w.sp, sizeof(hsp), hipMemcpyDeviceToHost, s));
  HIP_TRY(hipStreamSynchronize(s));
  *out_nmatches = (int64_t)hsp.cursor;
  if ((int64_t)hsp.cursor > out_cap)
    FAIL(GPUQ_ERR_OVERFLOW, "join: %lld matches exceed out_cap %lld",
         (long long)hsp.cursor, (long long)out_cap);
  return GPUQ_OK;
}

extern "C" int gpuq_join_probe_i64(void* stream, int64_t prows, gpuq_col pkey,
                                   const void* workspace, int64_t cap, int64_t brows,
                                   void* probe_workspace, int64_t probe_ws_bytes,
                                   uint32_t* out_p, uint32_t* out_b,
                                   int64_t out_cap, int64_t* out_nmatches) {
  return gpuq_join_probe_i64_typed(stream, prows, pkey, workspace, cap, brows,
                                   probe_workspace, probe_ws_bytes, 0,
                                   out_p, out_b, out_cap, out_nmatches);
}

/* ---- partitioned aggregation (mid/high cardinality) ----
 * The direct table is atomic-throughput-bound (~22 G f64-adds/s at 10M
 * groups, tools/diag_hash.py). Round-2 design: ONE non-stable 13-bit
 * bucket partition (8192 buckets; aggregation does not need stability, so
 * the two 16-bit ranked-scatter passes of round 1 — 2x the scatter
 * kernel's internal ceiling — are replaced by a histogram + reserve +
 * direct scatter whose per-block per-bucket runs are ~0.5 KB contiguous
 * stores), then per-chunk LDS-table aggregation: a 16 K-row chunk of
 * bucket-ordered data holds at most ~n_groups/8192 (+ one bucket boundary)
 * distinct keys, and a chunk that ever overflows the LDS table flushes it
 * to the global table mid-chunk and continues (skew-safe, no global
 * fallback). */

#define PAGG_BUCKETS_MAX 16384
#define PAGG_TILE (256 * 1024)    /* rows partitioned per block */

template <int BUCKETS>
DEV int pagg_bucket(int64_t k) {
  /* top bits below the sign of the 32-bit murmur */
  return (int)(((uint32_t)mm3_hash_long(k, 42) >> 12) & (BUCKETS - 1));
}

/* global bucket histogram; NULL-key rows count via key 0, real -1 keys via
 * their own hash — the scatter uses identical bucketing and marks those
 * records EMPTY, so layout and content agree. */
template <int BUCKETS>
__global__ void k_pagg_ghist(int64_t n, const int64_t* keys, const uint8_t* kvalid,
                             unsigned int* ghist) {
  __shared__ unsigned int h[BUCKETS];
  for (int b = threadIdx.x; b < BUCKETS; b += blockDim.x) h[b] = 0;
  __syncthreads();
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += gs) {
    int64_t k = bit_valid(kvalid, i) ? keys[i] : 0;
    atomicAdd(&h[pagg_bucket<BUCKETS>(k)], 1u);
  }
  __syncthreads();
  for (int b = threadIdx.x; b < BUCKETS; b += blockDim.x)
    if (h[b]) atomicAdd(&ghist[b], h[b]);
}

/* single-block exclusive scan of the bucket histogram -> running cursors */
template <int BUCKETS>
__global__ void k_pagg_scan(const unsigned int* ghist, unsigned int* gcursor) {
  __shared__ unsigned int buf[BUCKETS];
  for (int b = threadIdx.x; b < BUCKETS; b += blockDim.x)
    buf[b] = ghist[b];
  __syncthreads();
  if (threadIdx.x == 0) {           /* serial adds, once per aggregate call */
    unsigned int run = 0;
    for (int b = 0; b < BUCKETS; b++) {
      unsigned int t = buf[b];
      buf[b] = run;
      run += t;
    }
  }
  __syncthreads();
  for (int b = threadIdx.x; b < BUCKETS; b += blockDim.x)
    gcursor[b] = buf[b];
}

/* bucket scatter: per block, histogram its tile, reserve per-bucket ranges
 * with ONE global atomicAdd per (block, bucket), then write (key, val)
 * records bucket-contiguously. Special rows (NULL key / key == -1 ==
 * AGG_EMPTY) accumulate into the special slots and leave EMPTY records. */
template <int OPS, int BUCKETS, int SBLOCK, int ST /*0 plain,1 sc1,2 nt*/>
__global__ __launch_bounds__(SBLOCK)
void k_pagg_scatter(int64_t n, const int64_t* keys, const uint8_t* kvalid,
                    const double* vals, unsigned int* gcursor,
                    ulonglong2* recs, agg_special* sp) {
  __shared__ unsigned int h[BUCKETS];
  __shared__ unsigned int base[BUCKETS];
  const int64_t t0 = (int64_t)blockIdx.x * PAGG_TILE;
  const int64_t t1 = min(t0 + (int64_t)PAGG_TILE, n);
  for (int b = threadIdx.x; b < BUCKETS; b += blockDim.x) h[b] = 0;
  __syncthreads();
  for (int64_t i = t0 + threadIdx.x; i < t1; i += blockDim.x) {
    int64_t k = bit_valid(kvalid, i) ? keys[i] : 0;
    atomicAdd(&h[pagg_bucket<BUCKETS>(k)], 1u);
  }
  __syncthreads();
  for (int b = threadIdx.x; b < BUCKETS; b += blockDim.x)
    base[b] = h[b] ? atomicAdd(&gcursor[b], h[b]) : 0u;
  __syncthreads();
  for (int64_t i = t0 + threadIdx.x; i < t1; i += blockDim.x) {
    bool kv = bit_valid(kvalid, i);
    int64_t k = kv ? keys[i] : 0;
    double v = vals[i];
    unsigned int pos = atomicAdd(&base[pagg_bucket<BUCKETS>(k)], 1u);
    ulonglong2 rec;
    if (!kv || (unsigned long long)k == AGG_EMPTY) {
      unsigned long long* pseen = kv ? &sp->m1_seen : &sp->nul_seen;
      double* psum = kv ? &sp->m1_sum : &sp->nul_sum;
      unsigned long long* pcnt = kv ? &sp->m1_cnt : &sp->nul_cnt;
      atomicMax(pseen, 1ull);
      if (OPS & AGG_OP_SUM) atomicAdd(psum, v);
      if (OPS & AGG_OP_COUNT) atomicAdd(pcnt, 1ull);
      rec.x = AGG_EMPTY;
      rec.y = 0;
    } else {
      rec.x = (unsigned long long)k;
      rec.y = (unsigned long long)__double_as_longlong(v);
    }
    if (ST == 1) {
      /* write-through: the per-bucket runs are written temporally
       * scattered across the tile scan, so cached lines rarely collect a
       * full 64 B before eviction — skip the write-allocate RFO */
      __hip_atomic_store(&recs[pos].x, rec.x, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
      __hip_atomic_store(&recs[pos].y, rec.y, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
    } else if (ST == 2) {
      __builtin_nontemporal_store(rec.x, &recs[pos].x);
      __builtin_nontemporal_store(rec.y, &recs[pos].y);
    } else {
      recs[pos] = rec;
    }
  }
}

/* chunk aggregation over bucket-ordered records with mid-chunk flush: when
 * any thread cannot place its key (LDS table full — a skewed or
 * boundary-spanning chunk), the whole block merges the table into the
 * global one, clears it, and the failed inserts retry. */
template <int OPS, int SLOT, int SLOTS, int CHUNK, int CBLOCK>
__global__ __launch_bounds__(CBLOCK)
void k_agg_part(int64_t n, const ulonglong2* recs,
                unsigned long long* tab, agg_special* sp, int64_t cap_mask) {
  __shared__ unsigned long long lt[SLOTS * 3];
  const int64_t base = (int64_t)blockIdx.x * CHUNK;
  for (int j = threadIdx.x; j < SLOTS * 3; j += blockDim.x)
    lt[j] = (j % 3 == 0) ? AGG_EMPTY : 0;
  __syncthreads();
  auto merge_flush = [&]() {
    for (int j = threadIdx.x; j < SLOTS; j += blockDim.x) {
      unsigned long long k = lt[3 * j];
      if (k == AGG_EMPTY) continue;
      uint64_t slot = ((uint32_t)mm3_hash_long((int64_t)k, 42)) & (uint64_t)cap_mask;
      for (int64_t probes = 0;; probes++) {
        unsigned long long cur = __hip_atomic_load(&tab[SLOT * slot], __ATOMIC_RELAXED,
                                                   __HIP_MEMORY_SCOPE_AGENT);
        if (cur == k) break;
        if (cur == AGG_EMPTY) {
          unsigned long long prev = atomicCAS(&tab[SLOT * slot], AGG_EMPTY, k);
          if (prev == AGG_EMPTY || prev == k) break;
        }
        slot = (slot + 1) & (uint64_t)cap_mask;
        if (probes > cap_mask) { atomicMax(&sp->overflow, 1ull); return; }
      }
      if (OPS & AGG_OP_SUM)
        atomicAdd((double*)&tab[SLOT * slot + 1],
                  __longlong_as_double((long long)lt[3 * j + 1]));
      if (OPS & AGG_OP_COUNT) atomicAdd(&tab[SLOT * slot + 2], lt[3 * j + 2]);
    }
  };
  static_assert((CHUNK / CBLOCK) % 4 == 0, "round grouping");
  for (int r0 = 0; r0 < CHUNK / CBLOCK; r0 += 4) {
    /* batch 4 record loads in flight before the dependent hash->LDS-probe
     * chain (the LDS table path is latency-, not queue-bound) */
    ulonglong2 recv[4];
    bool havev[4];
    #pragma unroll
    for (int q = 0; q < 4; q++) {
      int64_t i = base + (int64_t)(r0 + q) * CBLOCK + threadIdx.x;
      havev[q] = i < n;
      if (havev[q]) recv[q] = recs[i];
    }
    #pragma unroll
    for (int q = 0; q < 4; q++)
      havev[q] = havev[q] && recv[q].x != AGG_EMPTY;
    for (int q = 0; q < 4; q++) {
    ulonglong2 rec = recv[q];
    bool have = havev[q];
    for (;;) {
      bool fail = false;
      if (have) {
        uint64_t slot = ((uint32_t)mm3_hash_long((int64_t)rec.x, 42)) &
                        (SLOTS - 1);
        int probes = 0;
        for (;; probes++) {
          unsigned long long cur = lt[3 * slot];
          if (cur == rec.x) break;
          if (cur == AGG_EMPTY) {
            unsigned long long prev = atomicCAS(&lt[3 * slot], AGG_EMPTY, rec.x);
            if (prev == AGG_EMPTY || prev == rec.x) break;
          }
          slot = (slot + 1) & (SLOTS - 1);
          if (probes >= SLOTS) { fail = true; break; }
        }
        if (!fail) {
          if (OPS & AGG_OP_SUM)
            atomicAdd((double*)&lt[3 * slot + 1],
                      __longlong_as_double((long long)rec.y));
          if (OPS & AGG_OP_COUNT) atomicAdd(&lt[3 * slot + 2], 1ull);
          have = false;
        }
      }
      if (__syncthreads_count(fail ? 1 : 0) == 0) break;
      merge_flush();
      __syncthreads();
      for (int j = threadIdx.x; j < SLOTS * 3; j += blockDim.x)
        lt[j] = (j % 3 == 0) ? AGG_EMPTY : 0;
      __syncthreads();
    }
    }
  }
  __syncthreads();
  merge_flush();
}

struct pagg_ws {
  ulonglong2* recs;
  unsigned int* ghist;
  unsigned int* gcursor;
  unsigned long long* tab;
  agg_special* sp;
};

static void pagg_ws_layout(int64_t n, int64_t cap, pagg_ws* w, char* base, int64_t* total) {
  int64_t off = 0;
  auto take = [&](int64_t bytes) {
    char* p = base ? base + off : nullptr;
    off += (bytes + 255) & ~255LL;
    return p;
  };
  w->recs = (ulonglong2*)take(n * 16);
  w->ghist = (unsigned int*)take(PAGG_BUCKETS_MAX * 4);
  w->gcursor = (unsigned int*)take(PAGG_BUCKETS_MAX * 4);
  w->tab = (unsigned long long*)take(cap * 24);
  w->sp = (agg_special*)take(sizeof(agg_special));
  *total = off;
}

extern "C" int64_t gpuq_hash_agg_part_workspace_bytes(int64_t n, int64_t cap) {
  pagg_ws w; int64_t total;
  pagg_ws_layout(n, cap, &w, nullptr, &total);
  return total;
}

/* Partitioned aggregation: same results contract as gpuq_hash_agg_i64_f64
 * (single-shot; requires non-null values). */
extern "C" int gpuq_hash_agg_partitioned(void* stream, int64_t n,
                                         gpuq_col key, gpuq_col val,
                                         void* workspace, int64_t cap, int32_t ops,
                                         int64_t* out_keys, uint8_t* out_key_valid,
                                         double* out_sums, uint8_t* out_sum_valid,
                                         int64_t* out_counts, int64_t* out_ngroups) {
  hipStream_t s = (hipStream_t)stream;
  if (cap <= 0 || (cap & (cap - 1)))
    FAIL(GPUQ_ERR_INVALID, "pagg: capacity %lld not a power of two", (long long)cap);
  if (n > 0xFFFFFFFELL)
    FAIL(GPUQ_ERR_INVALID, "pagg: nrows %lld > 2^32 (u32 bucket cursors)", (long long)n);
  if (key.dtype != GPUQ_INT64 || val.dtype != GPUQ_FLOAT64)
    FAIL(GPUQ_ERR_INVALID, "pagg: expected int64 key + float64 val");
  if (val.validity) FAIL(GPUQ_ERR_INVALID, "pagg: values must be non-null");
  pagg_ws w; int64_t need;
  pagg_ws_layout(n, cap, &w, (char*)workspace, &need);
  k_agg_init<3><<<grid1d(cap), 256, 0, s>>>(cap, w.tab);
  HIP_TRY(hipGetLastError());
  HIP_TRY(hipMemsetAsync(w.sp, 0, sizeof(agg_special), s));
  HIP_TRY(hipMemsetAsync(w.ghist, 0, PAGG_BUCKETS_MAX * 4, s));
  /* variant sweep knob: GPUQ_PAGG=0 (8192 buckets / 2048-slot 16K chunks)
   * 1 (16384 buckets / 4096-slot chunks) 2 (8192 buckets / 4096-slot) */
  static int variant = -1;
  if (variant < 0) {
    const char* e = getenv("GPUQ_PAGG");
    variant = e ? atoi(e) : 2;  /* 8192 buckets / 4096-slot chunks measured best */
  }
  if (n > 0) {
    { hipEvent_t _pe = prof_begin(s);
    if (variant == 1)
      k_pagg_ghist<16384><<<grid1d(n), 256, 0, s>>>(n, (const int64_t*)key.data,
                                                    key.validity, w.ghist);
    else
      k_pagg_ghist<8192><<<grid1d(n), 256, 0, s>>>(n, (const int64_t*)key.data,
                                                   key.validity, w.ghist);
    prof_end("pagg_ghist", s, _pe); }
    HIP_TRY(hipGetLastError());
    if (variant == 1)
      k_pagg_scan<16384><<<1, 256, 0, s>>>(w.ghist, w.gcursor);
    else
      k_pagg_scan<8192><<<1, 256, 0, s>>>(w.ghist, w.gcursor);
    HIP_TRY(hipGetLastError());
    int64_t nblocks = (n + PAGG_TILE - 1) / PAGG_TILE;
    static int st = -1;
    if (st < 0) {
      const char* e = getenv("GPUQ_PAGG_ST");
      st = e ? atoi(e) : 0;  /* plain stores measured best (sc1 +105%, nt +120%) */
    }
    { hipEvent_t _pe = prof_begin(s);
    #define PAGG_LAUNCH(OPS_, B_, T_) do { \
      if (st == 1) k_pagg_scatter<OPS_, B_, T_, 1><<<dim3((uint32_t)nblocks), T_, 0, s>>>( \
          n, (const int64_t*)key.data, key.validity, (const double*)val.data, \
          w.gcursor, w.recs, w.sp); \
      else if (st == 2) k_pagg_scatter<OPS_, B_, T_, 2><<<dim3((uint32_t)nblocks), T_, 0, s>>>( \
          n, (const int64_t*)key.data, key.validity, (const double*)val.data, \
          w.gcursor, w.recs, w.sp); \
      else k_pagg_scatter<OPS_, B_, T_, 0><<<dim3((uint32_t)nblocks), T_, 0, s>>>( \
          n, (const int64_t*)key.data, key.validity, (const double*)val.data, \
          w.gcursor, w.recs, w.sp); \
    } while (0)
    if (variant == 1) {
      if (ops == AGG_OP_SUM) PAGG_LAUNCH(AGG_OP_SUM, 16384, 1024);
      else PAGG_LAUNCH(3, 16384, 1024);
    } else {
      if (ops == AGG_OP_SUM) PAGG_LAUNCH(AGG_OP_SUM, 8192, 512);
      else PAGG_LAUNCH(3, 8192, 512);
    }
    #undef PAGG_LAUNCH
    prof_end("pagg_scatter", s, _pe); }
    HIP_TRY(hipGetLastError());
    { hipEvent_t _pe = prof_begin(s);
    int64_t nchunks = (n + 16384 - 1) / 16384;
    if (variant >= 1) {
      if (ops == AGG_OP_SUM)
        k_agg_part<AGG_OP_SUM, 3, 4096, 16384, 512><<<dim3((uint32_t)nchunks), 512, 0, s>>>(
            n, w.recs, w.tab, w.sp, cap - 1);
      else
        k_agg_part<3, 3, 4096, 16384, 512><<<dim3((uint32_t)nchunks), 512, 0, s>>>(
            n, w.recs, w.tab, w.sp, cap - 1);
    } else {
      if (ops == AGG_OP_SUM)
        k_agg_part<AGG_OP_SUM, 3, 2048, 16384, 256><<<dim3((uint32_t)nchunks), 256, 0, s>>>(
            n, w.recs, w.tab, w.sp, cap - 1);
      else
        k_agg_part<3, 3, 2048, 16384, 256><<<dim3((uint32_t)nchunks), 256, 0, s>>>(
            n, w.recs, w.tab, w.sp, cap - 1);
    }
    prof_end("pagg_chunks", s, _pe); }
    HIP_TRY(hipGetLastError());
  }
  agg_special hsp;
  HIP_TRY(hipMemcpyAsync(&hsp, w.sp, sizeof(hsp), hipMemcpyDeviceToHost, s));
  HIP_TRY(hipStreamSynchronize(s));
  if (hsp.overflow) FAIL(GPUQ_ERR_OVERFLOW, "pagg: table overflow (capacity %lld)", (long long)cap);
  dim3 cgrid((uint32_t)((cap + AGGC_CHUNK - 1) / AGGC_CHUNK));
  if (ops == AGG_OP_SUM)
    k_agg_compact<AGG_OP_SUM, 3><<<cgrid, 256, 0, s>>>(
        cap, w.tab, w.sp, out_keys, out_key_valid, out_sums, out_sum_valid, out_counts);
  else
    k_agg_compact<3, 3><<<cgrid, 256, 0, s>>>(
        cap, w.tab, w.sp, out_keys, out_key_valid, out_sums, out_sum_valid, out_counts);
  HIP_TRY(hipGetLastError());
  agg_special hsp2;
  HIP_TRY(hipMemcpyAsync(&hsp2, w.sp, sizeof(hsp2), hipMemcpyDeviceToHost, s));
  HIP_TRY(hipStreamSynchronize(s));
  *out_ngroups = (int64_t)hsp2.out_cursor;
  return GPUQ_OK;
}

/* ================= multi-aggregate ================= *//* ================= multi-aggregate ================= */
/*
 * One pass computing up to GPUQ_AGG_MAX_SPECS accumulators per group
 * (HashAggregateExec evaluates a LIST of aggregate expressions —
 * HashAggregateExec.scala:68-76; TPC-H Q1 has 8). Slot layout:
 * [key][acc_0]..[acc_{A-1}], stride = 1+A u64 words. SUM accs are f64
 * bits, COUNT accs u64. Small tables (cap*(1+A)*8 <= 64 KB) aggregate in
 * a per-block LDS table first (contention wall removal, same as the
 * single-value path). Round-1 constraint: SUM value columns must be
 * non-null (per-acc NULL tracking pairs a SUM with a COUNT spec).
 */

#define AGG_MAX_SPECS 12

struct agg_cols { const double* p[AGG_MAX_SPECS]; };
/* ops (Sum.scala:113-180, Count.scala, Min/Max.scala null-skipping update
 * expressions): 0=SUM_F64 1=COUNT(col) 2=COUNT(*) 3=SUM_I64(wrapping)
 * 4=MIN_I64 5=MAX_I64 6=MIN_F64 7=MAX_F64. MIN/MAX accumulators hold the
 * monotone u64 encoding (encode_i64/encode_f64) so u64 atomicMin/atomicMax
 * realize the reference ordering incl. NaN-greatest and -0.0 < 0.0; the
 * compact pass decodes. All value ops skip NULL rows; an all-NULL group's
 * acc stays at its init value — callers discriminate via a paired COUNT
 * (Sum/Min/Max evaluate to NULL iff no non-null input). */
struct agg_ops_spec { int op[AGG_MAX_SPECS]; };
struct agg_valid { const uint8_t* v[AGG_MAX_SPECS]; };
struct agg_outs { void* p[AGG_MAX_SPECS]; };

DEV unsigned long long aggm_acc_init(int op) {
  return (op == 4 || op == 6) ? ~0ULL : 0ULL;  /* MIN: u64 max; else 0 */
}

struct agg_multi_special {
  unsigned long long acc[2][AGG_MAX_SPECS];  /* [0]=key -1 group, [1]=NULL group */
  unsigned long long seen[2];
  unsigned long long out_cursor;
  unsigned long long overflow;
};

struct agg_multi_ws {
  unsigned long long* tab;
  agg_multi_special* sp;
};

static void agg_multi_ws_layout(int64_t cap, int nspecs, agg_multi_ws* w,
                                char* base, int64_t* total) {
  int64_t off = 0;
  auto take = [&](int64_t bytes) {
    char* p = base ? base + off : nullptr;
    off += (bytes + 255) & ~255LL;
    return p;
  };
  w->tab = (unsigned long long*)take(cap * 8 * (1 + nspecs));
  w->sp = (agg_multi_special*)take(sizeof(agg_multi_special));
  *total = off;
}

extern "C" int64_t gpuq_hash_agg_multi_workspace_bytes(int64_t cap, int32_t nspecs) {
  agg_multi_ws w; int64_t total;
  agg_multi_ws_layout(cap, nspecs, &w, nullptr, &total);
  return total;
}

__global__ void k_aggm_init(int64_t cap, int stride, unsigned long long* tab,
                            agg_ops_spec ops, agg_multi_special* sp) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; i < cap; i += gs) {
    tab[(int64_t)stride * i] = AGG_EMPTY;
    for (int j = 1; j < stride; j++)
      tab[(int64_t)stride * i + j] = aggm_acc_init(ops.op[j - 1]);
  }
  if (sp && blockIdx.x == 0 && threadIdx.x == 0) {
    sp->seen[0] = sp->seen[1] = 0;
    sp->out_cursor = 0;
    sp->overflow = 0;
    for (int w = 0; w < 2; w++)
      for (int j = 0; j < stride - 1; j++) sp->acc[w][j] = aggm_acc_init(ops.op[j]);
  }
}

DEV void aggm_update(unsigned long long* acc, int nspecs, const agg_ops_spec ops,
                     const agg_cols cols, const agg_valid av, int64_t i) {
  for (int j = 0; j < nspecs; j++) {
    int op = ops.op[j];
    if (op == 2) { atomicAdd(&acc[j], 1ull); continue; }  /* COUNT(*) */
    if (!bit_valid(av.v[j], i)) continue;  /* every other op skips NULLs */
    if (op == 0) {
      atomicAdd((double*)&acc[j], cols.p[j][i]);
    } else if (op == 1) {
      atomicAdd(&acc[j], 1ull);
    } else if (op == 3) {
      /* SUM(int64) -> int64 (Sum.scala resultType LongType; non-ansi
       * overflow wraps = two's-complement u64 add, bit-exact) */
      atomicAdd(&acc[j], (unsigned long long)((const int64_t*)cols.p[j])[i]);
    } else if (op == 4 || op == 5) {
      unsigned long long e = encode_i64(((const int64_t*)cols.p[j])[i]);
      if (op == 4) atomicMin(&acc[j], e); else atomicMax(&acc[j], e);
    } else {  /* 6/7: MIN/MAX f64 via the monotone encoding */
      unsigned long long e = encode_f64(cols.p[j][i]);
      if (op == 6) atomicMin(&acc[j], e); else atomicMax(&acc[j], e);
    }
  }
}

/* merge one accumulator word from a sub-table into the global table */
DEV void aggm_merge_acc(unsigned long long* dst, int op, unsigned long long v) {
  if (op == 0)
    atomicAdd((double*)dst, __longlong_as_double((long long)v));
  else if (op == 4 || op == 6) atomicMin(dst, v);
  else if (op == 5 || op == 7) atomicMax(dst, v);
  else atomicAdd(dst, v);  /* COUNT + SUM_I64 */
}

template <bool LDS>
__global__ __launch_bounds__(256)
void k_aggm_build(int64_t n, const int64_t* keys, const uint8_t* kvalid,
                  agg_cols cols, agg_ops_spec ops, agg_valid av, int nspecs,
                  unsigned long long* tab, agg_multi_special* sp,
                  int64_t cap_mask) {
  const int stride = 1 + nspecs;
  extern __shared__ __attribute__((aligned(16))) unsigned long long lt[];
  if (LDS) {
    for (int j = threadIdx.x; j < (int)(cap_mask + 1) * stride; j += blockDim.x) {
      int r = j % stride;
      lt[j] = (r == 0) ? AGG_EMPTY : aggm_acc_init(ops.op[r - 1]);
    }
    __syncthreads();
  }
  unsigned long long* t = LDS ? lt : tab;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += gs) {
    bool kv = bit_valid(kvalid, i);
    int64_t k = kv ? keys[i] : 0;
    if (!kv || (unsigned long long)k == AGG_EMPTY) {
      int which = kv ? 0 : 1;
      atomicMax(&sp->seen[which], 1ull);
      aggm_update(sp->acc[which], nspecs, ops, cols, av, i);
      continue;
    }
    uint64_t slot = ((uint32_t)mm3_hash_long(k, 42)) & (uint64_t)cap_mask;
    for (int probes = 0;; probes++) {
      unsigned long long cur = LDS ? t[(int64_t)stride * slot]
          : __hip_atomic_load(&t[(int64_t)stride * slot], __ATOMIC_RELAXED,
                              __HIP_MEMORY_SCOPE_AGENT);
      if (cur == (unsigned long long)k) break;
      if (cur == AGG_EMPTY) {
        unsigned long long prev = atomicCAS(&t[(int64_t)stride * slot], AGG_EMPTY,
                                            (unsigned long long)k);
        if (prev == AGG_EMPTY || prev == (unsigned long long)k) break;
      }
      slot = (slot + 1) & (uint64_t)cap_mask;
      if (probes > cap_mask) { atomicMax(&sp->overflow, 1ull); return; }
    }
    aggm_update(&t[(int64_t)stride * slot + 1], nspecs, ops, cols, av, i);
  }
  if (LDS) {
    __syncthreads();
    for (int sI = threadIdx.x; sI < (int)(cap_mask + 1); sI += blockDim.x) {
      unsigned long long k = lt[(int64_t)stride * sI];
      if (k == AGG_EMPTY) continue;
      uint64_t slot = ((uint32_t)mm3_hash_long((int64_t)k, 42)) & (uint64_t)cap_mask;
      for (;;) {
        unsigned long long cur = __hip_atomic_load(&tab[(int64_t)stride * slot],
                                                   __ATOMIC_RELAXED,
                                                   __HIP_MEMORY_SCOPE_AGENT);
        if (cur == k) break;
        if (cur == AGG_EMPTY) {
          unsigned long long prev = atomicCAS(&tab[(int64_t)stride * slot], AGG_EMPTY, k);
          if (prev == AGG_EMPTY || prev == k) break;
        }
        slot = (slot + 1) & (uint64_t)cap_mask;
      }
      for (int j = 0; j < nspecs; j++)
        aggm_merge_acc(&tab[(int64_t)stride * slot + 1 + j], ops.op[j],
                       lt[(int64_t)stride * sI + 1 + j]);
    }
  }
}

DEV void aggm_emit(void* out, int64_t o, int op, unsigned long long v) {
  if (op == 0) ((double*)out)[o] = __longlong_as_double((long long)v);
  else if (op == 4 || op == 5) ((int64_t*)out)[o] = decode_i64(v);
  else if (op == 6 || op == 7) ((double*)out)[o] = decode_f64(v);
  else ((int64_t*)out)[o] = (int64_t)v;
}

__global__ void k_aggm_compact(int64_t cap, const unsigned long long* tab,
                               agg_multi_special* sp, agg_ops_spec ops, int nspecs,
                               int64_t* out_keys, uint8_t* out_kvalid,
                               agg_outs outs) {
  constexpr int ROUNDS = AGGC_CHUNK / 256;
  const int stride = 1 + nspecs;
  __shared__ unsigned long long block_base;
  __shared__ uint32_t wtot[4];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int64_t base = (int64_t)blockIdx.x * AGGC_CHUNK;
  uint32_t occ_mask[ROUNDS / 32 + 1];
  for (int m = 0; m < ROUNDS / 32 + 1; m++) occ_mask[m] = 0;
  uint32_t lane_total = 0;
  for (int r = 0; r < ROUNDS; r++) {
    int64_t i = base + r * 256 + threadIdx.x;
    bool occ = i < cap && tab[(int64_t)stride * i] != AGG_EMPTY;
    if (occ) { occ_mask[r / 32] |= 1u << (r & 31); lane_total++; }
  }
  uint32_t incl = wave_inclusive_scan(lane_total);
  if (lane == WAVE - 1) wtot[wave] = incl;
  __syncthreads();
  if (threadIdx.x == 0) {
    uint32_t tot = 0;
    for (int w = 0; w < 4; w++) { uint32_t t = wtot[w]; wtot[w] = tot; tot += t; }
    block_base = tot ? atomicAdd(&sp->out_cursor, (unsigned long long)tot) : 0;
  }
  __syncthreads();
  int64_t o = (int64_t)block_base + wtot[wave] + (incl - lane_total);
  for (int r = 0; r < ROUNDS; r++) {
    if (!((occ_mask[r / 32] >> (r & 31)) & 1)) continue;
    int64_t i = base + r * 256 + threadIdx.x;
    out_keys[o] = (int64_t)tab[(int64_t)stride * i];
    out_kvalid[o] = 1;
    for (int j = 0; j < nspecs; j++)
      aggm_emit(outs.p[j], o, ops.op[j], tab[(int64_t)stride * i + 1 + j]);
    o++;
  }
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    for (int which = 0; which < 2; which++) {
      if (!sp->seen[which]) continue;
      int64_t q = (int64_t)atomicAdd(&sp->out_cursor, 1ull);
      out_keys[q] = which == 0 ? -1 : 0;
      out_kvalid[q] = which == 0 ? 1 : 0;
      for (int j = 0; j < nspecs; j++)
        aggm_emit(outs.p[j], q, ops.op[j], sp->acc[which][j]);
    }
  }
}

extern "C" int gpuq_hash_agg_multi(void* stream, int64_t n, gpuq_col key,
                                   const gpuq_col* vals, const int32_t* spec_ops,
                                   const int32_t* spec_cols, int32_t nspecs,
                                   void* workspace, int64_t cap,
                                   int32_t first_batch, int32_t finalize,
                                   int64_t* out_keys, uint8_t* out_key_valid,
                                   void* const* out_accs, int64_t* out_ngroups) {
  hipStream_t s = (hipStream_t)stream;
  if (cap <= 0 || (cap & (cap - 1)))
    FAIL(GPUQ_ERR_INVALID, "aggm: capacity %lld not a power of two", (long long)cap);
  if (nspecs < 1 || nspecs > AGG_MAX_SPECS)
    FAIL(GPUQ_ERR_INVALID, "aggm: nspecs %d not in [1,%d]", nspecs, AGG_MAX_SPECS);
  if (key.dtype != GPUQ_INT64) FAIL(GPUQ_ERR_INVALID, "aggm: key must be int64");
  agg_cols cols = {}; agg_ops_spec ops = {}; agg_valid av = {};
  for (int j = 0; j < nspecs; j++) {
    int op = spec_ops[j];
    ops.op[j] = op;
    if (op == 0 || op == 3 || (op >= 4 && op <= 7)) {
      const gpuq_col& c = vals[spec_cols[j]];
      int want = (op == 0 || op == 6 || op == 7) ? GPUQ_FLOAT64 : GPUQ_INT64;
      if (c.dtype != want) FAIL(GPUQ_ERR_INVALID, "aggm: value col dtype mismatch op %d", op);
      cols.p[j] = (const double*)c.data;
      av.v[j] = c.validity;  /* ops skip NULL rows; all-NULL groups stay at
                              * the init acc — callers pair a COUNT spec to
                              * emit SQL NULL (Sum/Min/Max.scala) */
    } else if (op == 1) {
      av.v[j] = vals[spec_cols[j]].validity;
    } else if (op != 2) {
      FAIL(GPUQ_ERR_INVALID, "aggm: bad op %d", op);
    }
  }
  int stride = 1 + nspecs;
  agg_multi_ws w; int64_t need;
  agg_multi_ws_layout(cap, nspecs, &w, (char*)workspace, &need);
  if (first_batch) {
    k_aggm_init<<<grid1d(cap), 256, 0, s>>>(cap, stride, w.tab, ops, w.sp);
    HIP_TRY(hipGetLastError());
  }
  if (n > 0) {
    int64_t lds_bytes = cap * stride * 8;
    bool lds = lds_bytes <= (64 << 10) && getenv("GPUQ_NO_LDS_AGG") == nullptr;
    { hipEvent_t _pe = prof_begin(s);
    if (lds)
      k_aggm_build<true><<<hash_grid(n), 256, (uint32_t)lds_bytes, s>>>(
          n, (const int64_t*)key.data, key.validity, cols, ops, av, nspecs,
          w.tab, w.sp, cap - 1);
    else
      k_aggm_build<false><<<hash_grid(n), 256, 0, s>>>(
          n, (const int64_t*)key.data, key.validity, cols, ops, av, nspecs,
          w.tab, w.sp, cap - 1);
    prof_end("aggm_build", s, _pe); }
    HIP_TRY(hipGetLastError());
  }
  if (finalize) {
    agg_multi_special hsp;
    HIP_TRY(hipMemcpyAsync(&hsp, w.sp, sizeof(hsp), hipMemcpyDeviceToHost, s));
    HIP_TRY(hipStreamSynchronize(s));
    if (hsp.overflow) FAIL(GPUQ_ERR_OVERFLOW, "aggm: hash table overflow (capacity %lld)", (long long)cap);
    agg_outs outs = {};
    for (int j = 0; j < nspecs; j++) outs.p[j] = out_accs[j];
    dim3 cgrid((uint32_t)((cap + AGGC_CHUNK - 1) / AGGC_CHUNK));
    k_aggm_compact<<<cgrid, 256, 0, s>>>(cap, w.tab, w.sp, ops, nspecs,
                                         out_keys, out_key_valid, outs);
    HIP_TRY(hipGetLastError());
    agg_multi_special hsp2;
    HIP_TRY(hipMemcpyAsync(&hsp2, w.sp, sizeof(hsp2), hipMemcpyDeviceToHost, s));
    HIP_TRY(hipStreamSynchronize(s));
    *out_ngroups = (int64_t)hsp2.out_cursor;
  }
  return GPUQ_OK;
}

/* ============ composite-key hash aggregate (multi-column GROUP BY) ======== */
/*
 * GROUP BY (k1..kK), K <= 4, int64 columns with independent NULLability —
 * the reference groups by an UnsafeRow of the key tuple
 * (UnsafeFixedWidthAggregationMap.java:39, TungstenAggregationIterator.scala:206).
 * Table layout: separate arrays — sig[cap] (u64 claim word), keys[K][cap],
 * kmask[cap] (bit c = key c non-NULL), acc[cap*nspecs]. A slot is claimed by
 * CAS(sig: EMPTY->PENDING), the tuple is published (keys + mask), then the
 * 64-bit signature is released into sig; probers spinning on PENDING retry
 * from the loop head (single attempt per iteration — wave64 lockstep-safe,
 * the publisher's branch always executes between iterations). Tuple equality
 * is verified against the published keys, so signature collisions only cost
 * extra probes, never correctness. No special slots needed: EMPTY/PENDING
 * are reserved signature values no tuple maps to (remapped), so the full
 * int64^K x NULL domain is exact.
 */

#define AGG_MAX_KEYS 4
#define AGGK_EMPTY 0ULL
#define AGGK_PENDING 1ULL

struct agg_keycols { const int64_t* k[AGG_MAX_KEYS]; const uint8_t* v[AGG_MAX_KEYS]; };
struct gpuq_outcols { void* p[AGG_MAX_KEYS]; };

struct aggk_ws {
  unsigned long long* sig;   /* [cap] */
  int64_t* keys;             /* [nkeys][cap] */
  uint8_t* kmask;            /* [cap] */
  unsigned long long* acc;   /* [cap][nspecs] */
  agg_multi_special* sp;     /* out_cursor/overflow only */
};

static void aggk_ws_layout(int64_t cap, int nkeys, int nspecs, aggk_ws* w,
                           char* base, int64_t* total) {
  int64_t off = 0;
  auto take = [&](int64_t bytes) {
    char* p = base ? base + off : nullptr;
    off += (bytes + 255) & ~255LL;
    return p;
  };
  w->sig = (unsigned long long*)take(cap * 8);
  w->keys = (int64_t*)take(cap * 8 * nkeys);
  w->kmask = (uint8_t*)take(cap);
  w->acc = (unsigned long long*)take(cap * 8 * nspecs);
  w->sp = (agg_multi_special*)take(sizeof(agg_multi_special));
  *total = off;
}

extern "C" int64_t gpuq_hash_agg_keys_workspace_bytes(int64_t cap, int32_t nkeys,
                                                      int32_t nspecs) {
  aggk_ws w; int64_t total;
  aggk_ws_layout(cap, nkeys, nspecs, &w, nullptr, &total);
  return total;
}

__global__ void k_aggk_init(int64_t cap, int nspecs, agg_ops_spec ops,
                            unsigned long long* sig, unsigned long long* acc,
                            agg_multi_special* sp) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; i < cap; i += gs) {
    sig[i] = AGGK_EMPTY;
    for (int j = 0; j < nspecs; j++)
      acc[(int64_t)nspecs * i + j] = aggm_acc_init(ops.op[j]);
  }
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    sp->out_cursor = 0;
    sp->overflow = 0;
  }
}

/* slot index: the reference's own seed-chained Murmur3 over the key tuple
 * (hash.scala:849-860 HashExpression.eval: NULL leaves the running hash
 * unchanged); signature: splitmix64 chain over (value|NULL-token) pairs. */
DEV void aggk_hash(const agg_keycols kc, int nkeys, int64_t i,
                   uint32_t* slot_hash, unsigned long long* sig, uint8_t* mask) {
  uint32_t h = 42;
  uint64_t s = 0x243F6A8885A308D3ULL;
  uint8_t m = 0;
  for (int c = 0; c < nkeys; c++) {
    bool kv = bit_valid(kc.v[c], i);
    int64_t k = kv ? kc.k[c][i] : 0;
    if (kv) { h = (uint32_t)mm3_hash_long(k, (int32_t)h); m |= (uint8_t)(1 << c); }
    s = splitmix64((s ^ (kv ? (uint64_t)k : 0xD1B54A32D192ED03ULL)) + (uint64_t)c);
  }
  if (s == AGGK_EMPTY || s == AGGK_PENDING) s = 2;
  *slot_hash = h; *sig = s; *mask = m;
}

__global__ __launch_bounds__(256)
void k_aggk_build(int64_t n, agg_keycols kc, int nkeys,
                  agg_cols cols, agg_ops_spec ops, agg_valid av, int nspecs,
                  unsigned long long* sig, int64_t* tkeys, uint8_t* tkmask,
                  unsigned long long* acc, agg_multi_special* sp,
                  int64_t cap, int64_t cap_mask) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += gs) {
    uint32_t h; unsigned long long mysig; uint8_t m;
    aggk_hash(kc, nkeys, i, &h, &mysig, &m);
    uint64_t slot = h & (uint64_t)cap_mask;
    int64_t probes = 0;
    for (;;) {
      unsigned long long cur = __hip_atomic_load(&sig[slot], __ATOMIC_ACQUIRE,
                                                 __HIP_MEMORY_SCOPE_AGENT);
      if (cur == AGGK_EMPTY) {
        unsigned long long prev = atomicCAS(&sig[slot], AGGK_EMPTY, AGGK_PENDING);
        if (prev == AGGK_EMPTY) {
          for (int c = 0; c < nkeys; c++)
            tkeys[(int64_t)c * cap + slot] =
                bit_valid(kc.v[c], i) ? kc.k[c][i] : 0;
          tkmask[slot] = m;
          __hip_atomic_store(&sig[slot], mysig, __ATOMIC_RELEASE,
                             __HIP_MEMORY_SCOPE_AGENT);
          break;
        }
        continue;  /* lost the claim: reload (sees PENDING or a sig) */
      }
      if (cur == AGGK_PENDING) continue;  /* publisher runs between iterations */
      if (cur == mysig) {
        bool eq = tkmask[slot] == m;
        for (int c = 0; eq && c < nkeys; c++)
          if ((m >> c) & 1)
            eq = tkeys[(int64_t)c * cap + slot] == kc.k[c][i];
        if (eq) break;
      }
      slot = (slot + 1) & (uint64_t)cap_mask;
      if (++probes > cap_mask) { atomicMax(&sp->overflow, 1ull); return; }
    }
    aggm_update(&acc[(int64_t)nspecs * slot], nspecs, ops, cols, av, i);
  }
}

__global__ void k_aggk_compact(int64_t cap, int nkeys, int nspecs,
                               const unsigned long long* sig, const int64_t* tkeys,
                               const uint8_t* tkmask, const unsigned long long* acc,
                               agg_multi_special* sp, agg_ops_spec ops,
                               gpuq_outcols okeys, uint8_t* out_kmask,
                               agg_outs outs) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  const int lane = threadIdx.x & (WAVE - 1);
  for (; i - lane < cap; i += gs) {   /* whole waves iterate together */
    bool occ = i < cap && sig[i] > AGGK_PENDING;
    uint64_t mask = __ballot(occ);
    if (!mask) continue;
    int cnt = __popcll(mask);
    unsigned long long base = 0;
    if (lane == __ffsll((unsigned long long)mask) - 1)
      base = atomicAdd(&sp->out_cursor, (unsigned long long)cnt);
    base = __shfl(base, __ffsll((unsigned long long)mask) - 1);
    if (!occ) continue;
    int64_t o = (int64_t)base + __popcll(mask & ((1ULL << lane) - 1));
    for (int c = 0; c < nkeys; c++)
      ((int64_t*)okeys.p[c])[o] = tkeys[(int64_t)c * cap + i];
    out_kmask[o] = tkmask[i];
    for (int j = 0; j < nspecs; j++)
      aggm_emit(outs.p[j], o, ops.op[j], acc[(int64_t)nspecs * i + j]);
  }
}

extern "C" int gpuq_hash_agg_keys(void* stream, int64_t n,
                                  const gpuq_col* key_cols, int32_t nkeys,
                                  const gpuq_col* vals, const int32_t* spec_ops,
                                  const int32_t* spec_cols, int32_t nspecs,
                                  void* workspace, int64_t cap,
                                  int32_t first_batch, int32_t finalize,
                                  void* const* out_keys, uint8_t* out_kmask,
                                  void* const* out_accs, int64_t* out_ngroups) {
  hipStream_t s = (hipStream_t)stream;
  if (cap <= 0 || (cap & (cap - 1)))
    FAIL(GPUQ_ERR_INVALID, "aggk: capacity %lld not a power of two", (long long)cap);
  if (nkeys < 1 || nkeys > AGG_MAX_KEYS)
    FAIL(GPUQ_ERR_INVALID, "aggk: nkeys %d not in [1,%d]", nkeys, AGG_MAX_KEYS);
  if (nspecs < 1 || nspecs > AGG_MAX_SPECS)
    FAIL(GPUQ_ERR_INVALID, "aggk: nspecs %d not in [1,%d]", nspecs, AGG_MAX_SPECS);
  agg_keycols kc = {};
  for (int c = 0; c < nkeys; c++) {
    if (key_cols[c].dtype != GPUQ_INT64)
      FAIL(GPUQ_ERR_INVALID, "aggk: key col %d must be int64", c);
    kc.k[c] = (const int64_t*)key_cols[c].data;
    kc.v[c] = key_cols[c].validity;
  }
  agg_cols cols = {}; agg_ops_spec ops = {}; agg_valid av = {};
  for (int j = 0; j < nspecs; j++) {
    int op = spec_ops[j];
    ops.op[j] = op;
    if (op == 0 || op == 3 || (op >= 4 && op <= 7)) {
      const gpuq_col& c = vals[spec_cols[j]];
      int want = (op == 0 || op == 6 || op == 7) ? GPUQ_FLOAT64 : GPUQ_INT64;
      if (c.dtype != want) FAIL(GPUQ_ERR_INVALID, "aggk: value col dtype mismatch op %d", op);
      cols.p[j] = (const double*)c.data;
      av.v[j] = c.validity;
    } else if (op == 1) {
      av.v[j] = vals[spec_cols[j]].validity;
    } else if (op != 2) {
      FAIL(GPUQ_ERR_INVALID, "aggk: bad op %d", op);
    }
  }
  aggk_ws w; int64_t need;
  aggk_ws_layout(cap, nkeys, nspecs, &w, (char*)workspace, &need);
  if (first_batch) {
    k_aggk_init<<<grid1d(cap), 256, 0, s>>>(cap, nspecs, ops, w.sig, w.acc, w.sp);
    HIP_TRY(hipGetLastError());
  }
  if (n > 0) {
    { hipEvent_t _pe = prof_begin(s);
    k_aggk_build<<<hash_grid(n), 256, 0, s>>>(
        n, kc, nkeys, cols, ops, av, nspecs, w.sig, w.keys, w.kmask, w.acc,
        w.sp, cap, cap - 1);
    prof_end("aggk_build", s, _pe); }
    HIP_TRY(hipGetLastError());
  }
  if (finalize) {
    agg_multi_special hsp;
    HIP_TRY(hipMemcpyAsync(&hsp, w.sp, sizeof(hsp), hipMemcpyDeviceToHost, s));
    HIP_TRY(hipStreamSynchronize(s));
    if (hsp.overflow) FAIL(GPUQ_ERR_OVERFLOW, "aggk: hash table overflow (capacity %lld)", (long long)cap);
    gpuq_outcols okeys = {};
    for (int c = 0; c < nkeys; c++) okeys.p[c] = (void*)out_keys[c];
    agg_outs outs = {};
    for (int j = 0; j < nspecs; j++) outs.p[j] = (void*)out_accs[j];
    k_aggk_compact<<<grid1d(cap), 256, 0, s>>>(cap, nkeys, nspecs, w.sig, w.keys,
                                               w.kmask, w.acc, w.sp, ops, okeys,
                                               out_kmask, outs);
    HIP_TRY(hipGetLastError());
    agg_multi_special hsp2;
    HIP_TRY(hipMemcpyAsync(&hsp2, w.sp, sizeof(hsp2), hipMemcpyDeviceToHost, s));
    HIP_TRY(hipStreamSynchronize(s));
    *out_ngroups = (int64_t)hsp2.out_cursor;
  }
  return GPUQ_OK;
}

/* ============ multi-column partition ids (seed-chained Murmur3) =========== */
/* pid = Pmod(Murmur3Hash(k1..kK, 42), n) with the hash chained column-wise
 * and NULL columns passing the running seed through unchanged —
 * HashPartitioning.partitionIdExpression (partitioning.scala:328) over
 * HashExpression.eval (hash.scala:849-860). */

__global__ void k_partition_pids_multi(int64_t n, agg_keycols kc, int nkeys,
                                       int32_t nparts, uint64_t* pid_as_key,
                                       uint32_t* idx,
                                       unsigned long long* counts) {
  __shared__ uint32_t h[256];
  bool lds_counts = nparts <= 256;
  if (lds_counts)
    for (int b = threadIdx.x; b < nparts; b += blockDim.x) h[b] = 0;
  __syncthreads();
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int32_t hsh = 42;
    for (int c = 0; c < nkeys; c++)
      if (bit_valid(kc.v[c], i)) hsh = mm3_hash_long(kc.k[c][i], hsh);
    int pid = spark_pmod(hsh, nparts);
    pid_as_key[i] = (uint64_t)pid;
    idx[i] = (uint32_t)i;
    if (lds_counts) atomicAdd(&h[pid], 1u);
    else atomicAdd(&counts[pid], 1ull);
  }
  __syncthreads();
  if (lds_counts)
    for (int b = threadIdx.x; b < nparts; b += blockDim.x)
      if (h[b]) atomicAdd(&counts[b], (unsigned long long)h[b]);
}

/* shared tail of the partition entry points: stable radix over the pid */
static int partition_scatter_tail(hipStream_t s, int64_t n, sort_ws& w,
                                  int32_t nparts, uint32_t* out_perm) {
  scatter_geom geom = get_sort_geom();
  int tile = geom.block * geom.items;
  int64_t nb = sort_nblocks(n, tile);
  int passes = nparts > 256 ? 2 : 1;
  uint64_t *kin = w.ka, *kout = w.kb;
  uint32_t *iin = w.ia, *iout = w.ib;
  for (int p = 0; p < passes; p++) {
    { hipEvent_t _pe = prof_begin(s);
    k_radix_hist<0><<<dim3((uint32_t)nb), 256, 0, s>>>(n, kin, p * 8, w.hist, (int)nb, tile);
    prof_end("radix_hist", s, _pe); }
    HIP_TRY(hipGetLastError());
    int rc = exclusive_scan_u32(s, (int64_t)256 * nb, w.hist, w.hist_scan, w.block_sums);
    if (rc) return rc;
    uint32_t* iout_pass = (p == passes - 1) ? out_perm : iout;
    { hipEvent_t _pe = prof_begin(s);
    launch_scatter<0, false>(s, geom, nb, n, kin, iin, kout, iout_pass, w.hist_scan,
                             p * 8, 0, nullptr, nullptr, nullptr, 0);
    prof_end("radix_scatter", s, _pe); }
    HIP_TRY(hipGetLastError());
    uint64_t* tk = kin; kin = kout; kout = tk;
    uint32_t* ti = iin; iin = iout_pass; iout = ti;
  }
  return GPUQ_OK;
}

extern "C" int gpuq_partition_perm_multi(void* stream, int64_t n,
                                         const gpuq_col* key_cols, int32_t nkeys,
                                         int32_t nparts, uint32_t* out_perm,
                                         int64_t* out_counts,
                                         void* workspace, int64_t workspace_bytes) {
  hipStream_t s = (hipStream_t)stream;
  if (n > 0xFFFFFFFFLL) FAIL(GPUQ_ERR_INVALID, "partition: nrows %lld > 2^32", (long long)n);
  if (nparts < 1 || nparts > 65536)
    FAIL(GPUQ_ERR_INVALID, "partition: num_parts %d not in [1,65536]", nparts);
  if (nkeys < 1 || nkeys > AGG_MAX_KEYS)
    FAIL(GPUQ_ERR_INVALID, "partition: nkeys %d not in [1,%d]", nkeys, AGG_MAX_KEYS);
  agg_keycols kc = {};
  for (int c = 0; c < nkeys; c++) {
    if (key_cols[c].dtype != GPUQ_INT64)
      FAIL(GPUQ_ERR_INVALID, "partition: key col %d must be int64", c);
    kc.k[c] = (const int64_t*)key_cols[c].data;
    kc.v[c] = key_cols[c].validity;
  }
  sort_ws w; int64_t need;
  sort_ws_layout(n, 256, &w, (char*)workspace, &need);
  if (workspace_bytes < need)
    FAIL(GPUQ_ERR_INVALID, "partition: workspace %lld < %lld", (long long)workspace_bytes, (long long)need);
  HIP_TRY(hipMemsetAsync(out_counts, 0, (size_t)nparts * 8, s));
  if (n == 0) return GPUQ_OK;
  { hipEvent_t _pe = prof_begin(s);
  k_partition_pids_multi<<<grid1d(n), 256, 0, s>>>(n, kc, nkeys, nparts, w.ka, w.ia,
                                                   (unsigned long long*)out_counts);
  prof_end("partition_pids", s, _pe); }
  HIP_TRY(hipGetLastError());
  return partition_scatter_tail(s, n, w, nparts, out_perm);
}

/* out[i] = col[perm[i]] with perm[i] == 0xFFFFFFFF (JOIN_NIL) producing a
 * NULL output row (outer-join build side); out_bits = resulting validity
 * (src validity AND not-NIL). */
__global__ void k_gather_nullable(int64_t n, const uint64_t* col,
                                  const uint8_t* src_valid,
                                  const uint32_t* perm, uint64_t* out,
                                  uint8_t* out_bits) {
  int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t nbytes = (n + 7) >> 3;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; b < nbytes; b += gs) {
    uint8_t v = 0;
    int64_t row0 = b << 3;
    int top = (int)((n - row0) < 8 ? (n - row0) : 8);
    for (int t = 0; t < top; t++) {
      uint32_t p = perm[row0 + t];
      bool ok = p != 0xFFFFFFFFu;
      out[row0 + t] = ok ? col[p] : 0;
      if (ok && (!src_valid || ((src_valid[p >> 3] >> (p & 7)) & 1)))
        v |= (uint8_t)(1 << t);
    }
    out_bits[b] = v;
  }
}

extern "C" int gpuq_gather_nullable(void* stream, int64_t nrows, gpuq_col col,
                                    const uint32_t* perm, void* out,
                                    uint8_t* out_bits) {
  hipStream_t s = (hipStream_t)stream;
  if (col.dtype != GPUQ_INT64 && col.dtype != GPUQ_FLOAT64)
    FAIL(GPUQ_ERR_INVALID, "gather_nullable: unsupported dtype %d", col.dtype);
  if (nrows == 0) return GPUQ_OK;
  int64_t nbytes = (nrows + 7) >> 3;
  k_gather_nullable<<<grid1d(nbytes), 256, 0, s>>>(
      nrows, (const uint64_t*)col.data, col.validity, perm, (uint64_t*)out,
      out_bits);
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

/* ================= validity-bitmap utilities ================= */
/* The ColumnVector contract carries NULLs everywhere (ColumnVector.java:
 * 58-366); these kernels move Arrow validity bitmaps (LSB-first) through
 * permutations and across the exchange (bitmaps travel as u8 columns in
 * the all-to-all because row split points are not byte-aligned). */

__global__ void k_gather_bits(int64_t n, const uint8_t* src, const uint32_t* perm,
                              uint8_t* out) {
  int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;   /* output byte */
  int64_t nbytes = (n + 7) >> 3;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; b < nbytes; b += gs) {
    uint8_t v = 0;
    int64_t row0 = b << 3;
    int top = (int)((n - row0) < 8 ? (n - row0) : 8);
    for (int t = 0; t < top; t++) {
      uint32_t p = perm[row0 + t];
      v |= (uint8_t)(((src[p >> 3] >> (p & 7)) & 1) << t);
    }
    out[b] = v;
  }
}

extern "C" int gpuq_gather_bits(void* stream, int64_t nrows, const uint8_t* src_bits,
                                const uint32_t* perm, uint8_t* out_bits) {
  hipStream_t s = (hipStream_t)stream;
  if (nrows == 0) return GPUQ_OK;
  int64_t nbytes = (nrows + 7) >> 3;
  k_gather_bits<<<grid1d(nbytes), 256, 0, s>>>(nrows, src_bits, perm, out_bits);
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

__global__ void k_bits_to_u8(int64_t n, const uint8_t* bits, uint8_t* out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += gs) out[i] = (bits[i >> 3] >> (i & 7)) & 1;
}

extern "C" int gpuq_bits_to_u8(void* stream, int64_t nrows, const uint8_t* bits,
                               uint8_t* out) {
  hipStream_t s = (hipStream_t)stream;
  if (nrows == 0) return GPUQ_OK;
  k_bits_to_u8<<<grid1d(nrows), 256, 0, s>>>(nrows, bits, out);
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

__global__ void k_u8_to_bits(int64_t n, const uint8_t* u8, uint8_t* bits) {
  int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t nbytes = (n + 7) >> 3;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; b < nbytes; b += gs) {
    uint8_t v = 0;
    int64_t row0 = b << 3;
    int top = (int)((n - row0) < 8 ? (n - row0) : 8);
    for (int t = 0; t < top; t++) v |= (uint8_t)((u8[row0 + t] & 1) << t);
    bits[b] = v;
  }
}

extern "C" int gpuq_u8_to_bits(void* stream, int64_t nrows, const uint8_t* u8,
                               uint8_t* out_bits) {
  hipStream_t s = (hipStream_t)stream;
  if (nrows == 0) return GPUQ_OK;
  int64_t nbytes = (nrows + 7) >> 3;
  k_u8_to_bits<<<grid1d(nbytes), 256, 0, s>>>(nrows, u8, out_bits);
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

__global__ void k_nonzero_to_bits(int64_t n, const int64_t* in, uint8_t* bits) {
  int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t nbytes = (n + 7) >> 3;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; b < nbytes; b += gs) {
    uint8_t v = 0;
    int64_t row0 = b << 3;
    int top = (int)((n - row0) < 8 ? (n - row0) : 8);
    for (int t = 0; t < top; t++) v |= (uint8_t)((in[row0 + t] != 0 ? 1 : 0) << t);
    bits[b] = v;
  }
}

/* validity bitmap from an int64 column: bit i = (in[i] != 0). Used to turn
 * merged partial COUNTs into the NULL-ness of merged SUM/MIN/MAX results
 * (Sum.scala: result is NULL iff no non-null input). */
extern "C" int gpuq_nonzero_to_bits(void* stream, int64_t nrows, const int64_t* in,
                                    uint8_t* out_bits) {
  hipStream_t s = (hipStream_t)stream;
  if (nrows == 0) return GPUQ_OK;
  int64_t nbytes = (nrows + 7) >> 3;
  k_nonzero_to_bits<<<grid1d(nbytes), 256, 0, s>>>(nrows, in, out_bits);
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

__global__ void k_maskbit_to_bits(int64_t n, const uint8_t* mask, int bit,
                                  uint8_t* bits) {
  int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t nbytes = (n + 7) >> 3;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; b < nbytes; b += gs) {
    uint8_t v = 0;
    int64_t row0 = b << 3;
    int top = (int)((n - row0) < 8 ? (n - row0) : 8);
    for (int t = 0; t < top; t++)
      v |= (uint8_t)(((mask[row0 + t] >> bit) & 1) << t);
    bits[b] = v;
  }
}

/* validity bitmap for key column `bit` from gpuq_hash_agg_keys' out_kmask */
extern "C" int gpuq_maskbit_to_bits(void* stream, int64_t nrows,
                                    const uint8_t* mask, int32_t bit,
                                    uint8_t* out_bits) {
  hipStream_t s = (hipStream_t)stream;
  if (bit < 0 || bit > 7) FAIL(GPUQ_ERR_INVALID, "maskbit: bit %d", bit);
  if (nrows == 0) return GPUQ_OK;
  int64_t nbytes = (nrows + 7) >> 3;
  k_maskbit_to_bits<<<grid1d(nbytes), 256, 0, s>>>(nrows, mask, bit, out_bits);
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

/* ================= min/max reduction (int64 column) ================= */
/* out_dev[0] = min (encoded u64), out_dev[1] = max (encoded), out_dev[2] =
 * count of valid rows. Used by the host packing rule (narrow composite keys
 * -> one i64) and for range sanity checks. */

__global__ void k_minmax_i64(int64_t n, const int64_t* data, const uint8_t* validity,
                             unsigned long long* out) {
  __shared__ unsigned long long smin[256], smax[256];
  __shared__ unsigned long long scnt;
  if (threadIdx.x == 0) scnt = 0;
  unsigned long long lmin = ~0ULL, lmax = 0, lcnt = 0;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += gs) {
    if (!bit_valid(validity, i)) continue;
    unsigned long long e = encode_i64(data[i]);
    if (e < lmin) lmin = e;
    if (e > lmax) lmax = e;
    lcnt++;
  }
  smin[threadIdx.x] = lmin; smax[threadIdx.x] = lmax;
  __syncthreads();
  if (lcnt) atomicAdd(&scnt, lcnt);
  for (int w = 128; w > 0; w >>= 1) {
    if (threadIdx.x < w) {
      if (smin[threadIdx.x + w] < smin[threadIdx.x]) smin[threadIdx.x] = smin[threadIdx.x + w];
      if (smax[threadIdx.x + w] > smax[threadIdx.x]) smax[threadIdx.x] = smax[threadIdx.x + w];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    atomicMin(&out[0], smin[0]);
    atomicMax(&out[1], smax[0]);
    if (scnt) atomicAdd(&out[2], scnt);
  }
}

extern "C" int gpuq_minmax_i64(void* stream, int64_t nrows, gpuq_col col,
                               unsigned long long* out_dev /* [3] */) {
  hipStream_t s = (hipStream_t)stream;
  if (col.dtype != GPUQ_INT64) FAIL(GPUQ_ERR_INVALID, "minmax: col must be int64");
  HIP_TRY(hipMemsetAsync(out_dev, 0xFF, 8, s));       /* min := u64 max */
  HIP_TRY(hipMemsetAsync(out_dev + 1, 0, 16, s));     /* max := 0, cnt := 0 */
  if (nrows == 0) return GPUQ_OK;
  k_minmax_i64<<<grid1d(nrows), 256, 0, s>>>(nrows, (const int64_t*)col.data,
                                             col.validity, out_dev);
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

/* ================= narrow-key pack/unpack ================= */
/* Pack two int64 key columns with known small ranges into one int64:
 * out = (a - a_bias) << shift | (b - b_bias). The host rule verifies
 * 0 <= a-a_bias < 2^(63-shift) and 0 <= b-b_bias < 2^shift via
 * gpuq_minmax_i64 first; the packed key feeds the fast single-key
 * aggregation path (LDS tables at low cardinality), then unpack restores
 * the tuple. NULLs must be handled by the caller (packing is only applied
 * to non-null key columns). */

__global__ void k_pack2_i64(int64_t n, const int64_t* a, const int64_t* b,
                            int64_t a_bias, int64_t b_bias, int shift,
                            int64_t* out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += gs)
    out[i] = ((a[i] - a_bias) << shift) | (b[i] - b_bias);
}

extern "C" int gpuq_pack2_i64(void* stream, int64_t nrows, const int64_t* a,
                              const int64_t* b, int64_t a_bias, int64_t b_bias,
                              int32_t shift, int64_t* out) {
  hipStream_t s = (hipStream_t)stream;
  if (shift < 1 || shift > 62) FAIL(GPUQ_ERR_INVALID, "pack2: shift %d", shift);
  if (nrows == 0) return GPUQ_OK;
  k_pack2_i64<<<grid1d(nrows), 256, 0, s>>>(nrows, a, b, a_bias, b_bias, shift, out);
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

__global__ void k_unpack2_i64(int64_t n, const int64_t* in, int64_t a_bias,
                              int64_t b_bias, int shift, int64_t* a, int64_t* b) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t gs = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += gs) {
    int64_t v = in[i];
    if (a) a[i] = (v >> shift) + a_bias;
    if (b) b[i] = (v & (((int64_t)1 << shift) - 1)) + b_bias;
  }
}

extern "C" int gpuq_unpack2_i64(void* stream, int64_t nrows, const int64_t* in,
                                int64_t a_bias, int64_t b_bias, int32_t shift,
                                int64_t* out_a, int64_t* out_b) {
  hipStream_t s = (hipStream_t)stream;
  if (shift < 1 || shift > 62) FAIL(GPUQ_ERR_INVALID, "unpack2: shift %d", shift);
  if (nrows == 0) return GPUQ_OK;
  k_unpack2_i64<<<grid1d(nrows), 256, 0, s>>>(nrows, in, a_bias, b_bias, shift,
                                              out_a, out_b);
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

/* ================= filter / project (SURVEY (f).2) ================= */
/*
 * FILTER — replaces FilterExec for single-comparison predicates
 * (col OP literal): stable compaction via the same ranked-scatter machinery
 * as the sort's validity split (pass=group 0, fail=group 1; WHERE keeps
 * only TRUE — a NULL comparison result drops the row, SQL 3VL).
 * PROJECT — replaces ProjectExec for elementwise binary arithmetic
 * (col OP col / col OP literal) on int64/float64.
 */

DEV bool filter_cmp_f64(double v, int op, double lit) {
  switch (op) {
    case 0: return v == lit;
    case 1: return v < lit;
    case 2: return v <= lit;
    case 3: return v > lit;
    case 4: return v >= lit;
    default: return v != lit;
  }
}
DEV bool filter_cmp_i64(int64_t v, int op, int64_t lit) {
  switch (op) {
    case 0: return v == lit;
    case 1: return v < lit;
    case 2: return v <= lit;
    case 3: return v > lit;
    case 4: return v >= lit;
    default: return v != lit;
  }
}

template <int DTYPE>
__global__ void k_filter_pred(int64_t n, const void* data, const uint8_t* validity,
                              int op, double lit_f, int64_t lit_i,
                              uint64_t* pid_as_key, uint32_t* idx,
                              unsigned long long* pass_count) {
  __shared__ unsigned int h;
  if (threadIdx.x == 0) h = 0;
  __syncthreads();
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  unsigned int mine = 0;
  for (; i < n; i += stride) {
    bool pass = bit_valid(validity, i);
    if (pass) {
      if (DTYPE == GPUQ_FLOAT64) pass = filter_cmp_f64(((const double*)data)[i], op, lit_f);
      else pass = filter_cmp_i64(((const int64_t*)data)[i], op, lit_i);
    }
    pid_as_key[i] = pass ? 0ull : 1ull;
    idx[i] = (uint32_t)i;
    if (pass) mine++;
  }
  if (mine) atomicAdd(&h, mine);
  __syncthreads();
  if (threadIdx.x == 0 && h) atomicAdd(pass_count, (unsigned long long)h);
}

extern "C" int64_t gpuq_filter_workspace_bytes(int64_t n) {
  return gpuq_sort_workspace_bytes(n);
}

extern "C" int gpuq_filter_cmp(void* stream, int64_t n, gpuq_col col, int32_t op,
                               double lit_f, int64_t lit_i,
                               uint32_t* out_perm, int64_t* out_count,
                               void* workspace, int64_t workspace_bytes) {
  hipStream_t s = (hipStream_t)stream;
  if (n > 0xFFFFFFFFLL) FAIL(GPUQ_ERR_INVALID, "filter: nrows %lld > 2^32", (long long)n);
  if (op < 0 || op > 5) FAIL(GPUQ_ERR_INVALID, "filter: bad op %d", op);
  if (col.dtype != GPUQ_INT64 && col.dtype != GPUQ_FLOAT64)
    FAIL(GPUQ_ERR_INVALID, "filter: unsupported dtype %d", col.dtype);
  sort_ws w; int64_t need;
  sort_ws_layout(n, 256, &w, (char*)workspace, &need);
  if (workspace_bytes < need)
    FAIL(GPUQ_ERR_INVALID, "filter: workspace %lld < %lld", (long long)workspace_bytes, (long long)need);
  HIP_TRY(hipMemsetAsync(out_count, 0, 8, s));
  if (n == 0) return GPUQ_OK;
  scatter_geom geom = get_sort_geom();
  int tile = geom.block * geom.items;
  int64_t nb = sort_nblocks(n, tile);
  { hipEvent_t _pe = prof_begin(s);
  if (col.dtype == GPUQ_FLOAT64)
    k_filter_pred<GPUQ_FLOAT64><<<grid1d(n), 256, 0, s>>>(
        n, col.data, col.validity, op, lit_f, lit_i, w.ka, w.ia,
        (unsigned long long*)out_count);
  else
    k_filter_pred<GPUQ_INT64><<<grid1d(n), 256, 0, s>>>(
        n, col.data, col.validity, op, lit_f, lit_i, w.ka, w.ia,
        (unsigned long long*)out_count);
  prof_end("filter_pred", s, _pe); }
  HIP_TRY(hipGetLastError());
  k_radix_hist<0><<<dim3((uint32_t)nb), 256, 0, s>>>(n, w.ka, 0, w.hist, (int)nb, tile);
  HIP_TRY(hipGetLastError());
  int rc = exclusive_scan_u32(s, 256 * nb, w.hist, w.hist_scan, w.block_sums);
  if (rc) return rc;
  { hipEvent_t _pe = prof_begin(s);
  launch_scatter<0, false>(s, geom, nb, n, w.ka, w.ia, w.kb, w.ib, w.hist_scan,
                           0, 0, nullptr, nullptr, nullptr, 0);
  prof_end("filter_scatter", s, _pe); }
  HIP_TRY(hipGetLastError());
  HIP_TRY(hipMemcpyAsync(out_perm, w.ib, n * 4, hipMemcpyDeviceToDevice, s));
  return GPUQ_OK;
}

/* out = a OP b (b = column or broadcast literal) */
template <int DTYPE, int OP, bool B_IS_LIT>
__global__ void k_project_binop(int64_t n, const void* a, const void* b,
                                double lit_f, int64_t lit_i, void* out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    if (DTYPE == GPUQ_FLOAT64) {
      double x = ((const double*)a)[i];
      double y = B_IS_LIT ? lit_f : ((const double*)b)[i];
      double r = OP == 0 ? x + y : OP == 1 ? x - y : OP == 2 ? x * y
                 : OP == 4 ? y - x : x / y;
      ((double*)out)[i] = r;
    } else {
      int64_t x = ((const int64_t*)a)[i];
      int64_t y = B_IS_LIT ? lit_i : ((const int64_t*)b)[i];
      int64_t r = OP == 0 ? x + y : OP == 1 ? x - y : OP == 2 ? x * y
                  : OP == 4 ? y - x : (y == 0 ? 0 : x / y);
      ((int64_t*)out)[i] = r;
    }
  }
}

extern "C" int gpuq_project_binop(void* stream, int64_t n, gpuq_col a,
                                  const void* b /* col data or NULL */,
                                  double lit_f, int64_t lit_i, int32_t op,
                                  void* out) {
  hipStream_t s = (hipStream_t)stream;
  if (op < 0 || op > 4) FAIL(GPUQ_ERR_INVALID, "project: bad op %d", op);
  if (a.validity) FAIL(GPUQ_ERR_INVALID, "project: validity not yet supported");
  dim3 g = grid1d(n);
#define PJ(DT, OPV) do { \
    if (b) k_project_binop<DT, OPV, false><<<g, 256, 0, s>>>(n, a.data, b, lit_f, lit_i, out); \
    else   k_project_binop<DT, OPV, true><<<g, 256, 0, s>>>(n, a.data, b, lit_f, lit_i, out); \
  } while (0)
  if (a.dtype == GPUQ_FLOAT64) {
    if (op == 0) PJ(GPUQ_FLOAT64, 0); else if (op == 1) PJ(GPUQ_FLOAT64, 1);
    else if (op == 2) PJ(GPUQ_FLOAT64, 2); else if (op == 4) PJ(GPUQ_FLOAT64, 4);
    else PJ(GPUQ_FLOAT64, 3);
  } else if (a.dtype == GPUQ_INT64) {
    if (op == 0) PJ(GPUQ_INT64, 0); else if (op == 1) PJ(GPUQ_INT64, 1);
    else if (op == 2) PJ(GPUQ_INT64, 2); else if (op == 4) PJ(GPUQ_INT64, 4);
    else PJ(GPUQ_INT64, 3);
  } else {
    FAIL(GPUQ_ERR_INVALID, "project: unsupported dtype %d", a.dtype);
  }
#undef PJ
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

/* int64 -> float64 cast (AVG = SUM/COUNT evaluation; Average.scala
 * evaluateExpression divides sum by count cast to double) */
__global__ void k_cast_i64_f64(int64_t n, const int64_t* in, double* out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) out[i] = (double)in[i];
}

extern "C" int gpuq_cast_i64_f64(void* stream, int64_t n, const int64_t* in,
                                 double* out) {
  k_cast_i64_f64<<<grid1d(n), 256, 0, (hipStream_t)stream>>>(n, in, out);
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}

/* RangeExec scan feed (basicPhysicalOperators.scala:630): id = start + i*step */
__global__ void k_range_i64(int64_t n, int64_t start, int64_t step, int64_t* out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) out[i] = start + i * step;
}

extern "C" int gpuq_range_i64(void* stream, int64_t n, int64_t start,
                              int64_t step, int64_t* out) {
  k_range_i64<<<grid1d(n), 256, 0, (hipStream_t)stream>>>(n, start, step, out);
  HIP_TRY(hipGetLastError());
  return GPUQ_OK;
}
