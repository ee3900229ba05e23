// CDNA4 (gfx950 / MI355X) kernels for the columnar keyed-stream fast path.
//
// Replaces the per-item Python/Rust hot loops of the reference
// (reference src/operators.rs stateful_batch + windowing.py fold logic)
// with single-pass device kernels over columnar event batches:
//
//   * k_window_agg_insert: fused {window-id computation, open-address
//     hash insert into HBM-resident keyed window state, watermark
//     (max-timestamp) tracking} — one read of the event batch.
//   * k_close_extract: scan the table, emit (key, window, value)
//     triples for windows below the close horizon, compacted via
//     wave-ballot + one atomic per wave.
//   * k_bucket_hist / k_bucket_scatter: key-hash bucketing producing
//     per-destination contiguous segments for the RCCL all-to-allv
//     exchange across workers-as-GPUs.
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//   - wave width 64 everywhere; ballots are 64-bit.
//   - memory-bound kernels: grid-stride loops, ≤2048 blocks.
//   - atomics are device-scope by default on CDNA, which is what we
//     need since per-XCD L2s are not coherent; the insert loop only
//     trusts plain loads for values that can never change once set
//     (slot keys are write-once between clears).
//   - optional wave-level duplicate aggregation (`DEDUP`) cuts atomic
//     contention when key cardinality is low (e.g. 2-key
//     benchmark_windowing workload).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <pybind11/stl.h>
#include <ATen/hip/HIPContext.h>

#include <chrono>
#include <type_traits>
#include <cstdlib>
#include <cstdint>
#include <cstring>
#include <vector>

#define WAVE 64
#define EMPTY_SLOT 0xFFFFFFFFFFFFFFFFULL

#define HIP_CHECK(expr)                                               \
  do {                                                                \
    hipError_t _e = (expr);                                           \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e)); \
  } while (0)

static inline int n_blocks(int64_t n, int block) {
  int64_t b = (n + block - 1) / block;
  if (b > 2048) b = 2048;
  if (b < 1) b = 1;
  return (int)b;
}

// Scatter-variant selection for the radix window path (A/B-able via
// env until the measured default settles; see profiles/).
enum ScatterKind { SCAT_FIXED = 0, SCAT_DIRECT = 1, SCAT_STAGED = 2 };

static ScatterKind scatter_kind_env() {
  if (const char* s = std::getenv("BYTEWAX_SCATTER")) {
    if (s[0] == 'd') return SCAT_DIRECT;
    if (s[0] == 'f') return SCAT_FIXED;
    if (s[0] == 's') return SCAT_STAGED;
  }
  if (const char* sd = std::getenv("BYTEWAX_SCATTER_DIRECT")) {
    if (atoi(sd) != 0) return SCAT_DIRECT;
  }
  return SCAT_STAGED;
}

// Experimental coalesced-placement staged variant ("staged2"): only
// consulted where it is implemented (COUNT, tumbling).
static bool scatter_staged2_env() {
  const char* s = std::getenv("BYTEWAX_SCATTER");
  return s != nullptr && std::strncmp(s, "staged2", 7) == 0;
}

// LDS-deduped region join (BYTEWAX_JOIN_LDS=0 falls back to the
// per-event region kernel for A/B).
static bool join_lds_env() {
  const char* s = std::getenv("BYTEWAX_JOIN_LDS");
  return s == nullptr || s[0] != '0';
}

// Extra segment coarsening: scatter segments cover 2^coarse table
// regions each (longer runs per segment; the agg kernel stages
// 2^(region_bits+coarse) LDS slots per block).
static int scatter_coarse_bits(ScatterKind kind) {
  if (const char* cb = std::getenv("BYTEWAX_SCATTER_COARSE_BITS")) {
    int v = atoi(cb);
    if (v >= 0 && v <= 4) return v;
  }
  return kind == SCAT_STAGED ? 2 : 0;
}

__device__ __forceinline__ uint64_t mix64(uint64_t x) {
  // splitmix64 finalizer.
  x ^= x >> 33;
  x *= 0xff51afd7ed558ccdULL;
  x ^= x >> 33;
  x *= 0xc4ceb9fe1a85ec53ULL;
  x ^= x >> 33;
  return x;
}

// Table slot addressing.  Two layouts share one code path:
//   region_bits == 0: classic whole-table linear probing.
//   region_bits  > 0: the table is partitioned into `nslots >>
//     region_bits` contiguous *regions*; a key's region comes from the
//     low hash bits and probing stays inside the region.  This is the
//     layout the radix-partitioned LDS aggregation path flushes into
//     (one workgroup owns one region → L2-local updates).
__device__ __forceinline__ uint64_t region_of(
    uint64_t h64, uint64_t mask, int region_bits) {
  return (h64 & mask) >> region_bits;
}

// Insert `inc` into the open-address table for `packed`, creating the
// slot if needed.  Table size is a power of two (`mask = nslots-1`).
// Returns false if the table (region) is full.
__device__ __forceinline__ bool hash_add(
    uint64_t* __restrict__ tkeys,
    unsigned long long* __restrict__ tvals,
    uint64_t mask,
    int region_bits,
    uint64_t packed,
    unsigned long long inc) {
  uint64_t h64 = mix64(packed);
  uint64_t h, probe_mask, base;
  if (region_bits == 0) {
    base = 0;
    probe_mask = mask;
    h = h64 & mask;
  } else {
    base = region_of(h64, mask, region_bits) << region_bits;
    probe_mask = (1ULL << region_bits) - 1;
    h = (h64 >> 32) & probe_mask;
  }
  for (uint64_t probes = 0; probes <= probe_mask; ++probes) {
    uint64_t slot = base | h;
    uint64_t cur = tkeys[slot];
    if (cur == packed) {
      atomicAdd(&tvals[slot], inc);
      return true;
    }
    if (cur == EMPTY_SLOT) {
      // Plain load may be stale across XCDs; the CAS is the truth.
      uint64_t prev = atomicCAS(
          (unsigned long long*)&tkeys[slot], EMPTY_SLOT, packed);
      if (prev == EMPTY_SLOT || prev == packed) {
        atomicAdd(&tvals[slot], inc);
        return true;
      }
      // Someone else claimed the slot with a different key; keep
      // probing.
    }
    h = (h + 1) & probe_mask;
  }
  return false;
}

__global__ void k_bump(int64_t* p, int64_t v) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *p += v;
}

// Aggregation modes baked at compile time per kernel instantiation.
enum AggMode { AGG_COUNT = 0, AGG_SUM = 1,
               // Scatter-only mode for the fused session path:
               // packed = key (no window), value = absolute ts.
               AGG_TS = 2 };

__device__ __forceinline__ uint64_t find_slot(
    uint64_t* __restrict__ tkeys, uint64_t mask, uint64_t packed);
__device__ __forceinline__ uint64_t find_slot_r(
    uint64_t* __restrict__ tkeys, uint64_t mask, int region_bits,
    uint64_t packed);

template <int MODE, bool DEDUP, typename TS = int64_t>
__global__ void k_window_agg_insert(
    const int32_t* __restrict__ keys,
    const TS* __restrict__ ts,
    const int64_t* __restrict__ vals,  // nullptr for COUNT
    int64_t n,
    uint64_t* __restrict__ tkeys,
    unsigned long long* __restrict__ tvals,
    uint64_t mask,
    int64_t align_ms,
    int64_t len_ms,
    int64_t off_ms,  // window stride; == len_ms for tumbling, < len_ms
                     // for sliding (an event lands in ceil(len/off)
                     // windows)
    int64_t ts_base,  // added to every timestamp (columnar sources can
                      // reuse one template batch across steps)
    int region_bits,  // table layout (see hash_add)
    unsigned long long* __restrict__ max_ts,  // device scalar (atomicMax)
    int* __restrict__ error_flag,
    const int64_t* __restrict__ ts_base_dev) {  // extra device-side
                      // offset, lets a captured hipGraph be replayed
                      // with advancing timestamps (see k_bump)
  if (ts_base_dev != nullptr) ts_base += *ts_base_dev;
  int lane = threadIdx.x & (WAVE - 1);
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  int64_t local_max = 0;
  // Wave-uniform iteration: every lane of a wave runs every loop
  // iteration (masked by `valid`) so cross-lane ops in the DEDUP path
  // never read from exited lanes.
  int64_t first = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  for (int64_t i = first; i - lane < n; i += stride) {
    bool valid = i < n;
    int64_t t = valid ? ((int64_t)ts[i] + ts_base) : 0;
    if (t > local_max) local_max = t;
    uint64_t packed = 0;
    unsigned long long inc = 0;
    if (valid) {
      int64_t win = (t - align_ms) / off_ms;
      packed = ((uint64_t)(uint32_t)(int32_t)win << 32) | (uint32_t)keys[i];
      inc = (MODE == AGG_COUNT) ? 1ULL : (unsigned long long)vals[i];
      if (off_ms < len_ms) {
        // Sliding: also insert into the earlier windows this event
        // overlaps ([win_lo, win)).  The newest window is handled by
        // the shared code below so the DEDUP wave-aggregation path
        // stays correct for tumbling.  Floor division (C trunc would
        // drop the earliest window for events within `len` of align).
        int64_t num = t - align_ms - len_ms;
        int64_t win_lo = num / off_ms;
        if (num % off_ms != 0 && num < 0) win_lo -= 1;
        win_lo += 1;
        for (int64_t wn = win_lo; wn < win; ++wn) {
          uint64_t p2 =
              ((uint64_t)(uint32_t)(int32_t)wn << 32) | (uint32_t)keys[i];
          if (!hash_add(tkeys, tvals, mask, region_bits, p2, inc)) {
            atomicExch(error_flag, 1);
          }
        }
      }
    }

    bool ok = true;
    if (DEDUP) {
      // Wave-aggregate duplicate (key, window) pairs: one atomic per
      // distinct pair per wave.  Worth it only for low-cardinality
      // keys; high-cardinality batches use the plain path.
      unsigned long long remaining = __ballot(valid);
      bool leader = false;
      unsigned long long agg = 0;
      while (remaining) {
        int l = __ffsll((unsigned long long)remaining) - 1;
        uint64_t lk = (uint64_t)__shfl((long long)packed, l);
        unsigned long long match =
            __ballot(valid && packed == lk);
        if (MODE == AGG_COUNT) {
          if (lane == l) {
            leader = true;
            agg = __popcll(match);
          }
        } else {
          // Segmented sum over matching lanes: XOR butterfly gives
          // every lane (incl. the leader) the full total.
          unsigned long long contrib =
              (valid && packed == lk) ? inc : 0ULL;
          for (int off = WAVE / 2; off > 0; off >>= 1) {
            contrib += (unsigned long long)__shfl_xor(
                (long long)contrib, off);
          }
          if (lane == l) {
            leader = true;
            agg = contrib;
          }
        }
        remaining &= ~match;
      }
      if (leader) ok = hash_add(tkeys, tvals, mask, region_bits, packed, agg);
    } else if (valid) {
      ok = hash_add(tkeys, tvals, mask, region_bits, packed, inc);
    }
    if (!ok) atomicExch(error_flag, 1);
  }
  // Wave-reduce the max timestamp, one atomic per wave.
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    int64_t other = __shfl_down((long long)local_max, off);
    if (other > local_max) local_max = other;
  }
  if ((threadIdx.x & (WAVE - 1)) == 0 && local_max > 0) {
    atomicMax(max_ts, (unsigned long long)local_max);
  }
}

// ---------------------------------------------------------------------------
// Radix-partitioned LDS-staged aggregation (the high-cardinality fast
// path).  Instead of one random device-scope atomic per event into an
// L2-missing table, events are first partitioned into contiguous
// per-region segments (two streaming passes), then one workgroup
// aggregates each region's events in an LDS-resident open-address
// table (LDS atomics) and flushes distinct keys once into the region's
// contiguous slice of the HBM table — turning ~N random global
// atomics into ~distinct-keys L2-local ones.
// ---------------------------------------------------------------------------

// ---------------------------------------------------------------------------
// Radix v2 (experimental, COUNT mode): two-level scheme that writes
// full 64-byte lines.
//
//   Pass A (k_scatter_coarse): events scatter into 256 *coarse*
//   buckets, staged through LDS in block-synchronous tiles so every
//   global write is one aligned 64 B line (8 packed events); partial
//   groups at block end go to a per-bucket residual area (unaligned,
//   ~5% of events).
//
//   Pass B (k_agg_cached): per coarse-bucket slice, aggregate through
//   a 4096-entry LDS cache; collisions and the final flush go through
//   hash_add into the bucket's contiguous span of table regions
//   (cache-local by construction).
// ---------------------------------------------------------------------------

#define V2_COARSE 256
#define V2_STAGE 32  // staged events per coarse bucket per tile

__global__ __launch_bounds__(512) void k_scatter_coarse(
    const int32_t* __restrict__ keys,
    const int64_t* __restrict__ ts,
    int64_t n,
    int64_t align_ms,
    int64_t len_ms,
    int64_t ts_base,
    uint64_t mask,
    int region_bits,
    int64_t cap_c,      // aligned-area capacity per coarse bucket (mult of 8)
    int64_t res_cap,    // residual-area capacity per coarse bucket
    int* __restrict__ gcur,     // [V2_COARSE] aligned-area cursors
    int* __restrict__ gres,     // [V2_COARSE] residual-area cursors
    uint64_t* __restrict__ ev,      // [V2_COARSE * cap_c]
    uint64_t* __restrict__ ev_res,  // [V2_COARSE * res_cap]
    unsigned long long* __restrict__ max_ts,
    int* __restrict__ error_flag) {
  __shared__ uint64_t stage[V2_COARSE][V2_STAGE];
  __shared__ int lcnt[V2_COARSE];
  int nb = (int)(((mask + 1) >> region_bits));
  int shift = 0;  // region index -> coarse bucket shift
  while ((nb >> shift) > V2_COARSE) ++shift;
  for (int b = threadIdx.x; b < V2_COARSE; b += blockDim.x) lcnt[b] = 0;
  __syncthreads();

  const int TILE = 8;  // events per thread per tile
  int64_t chunk = (int64_t)blockDim.x * TILE;
  int64_t start = (int64_t)blockIdx.x * chunk;
  int64_t gstride = (int64_t)gridDim.x * chunk;
  int64_t local_max = 0;
  for (int64_t tile0 = start; tile0 < n; tile0 += gstride) {
    for (int t = 0; t < TILE; ++t) {
      int64_t i = tile0 + (int64_t)t * blockDim.x + threadIdx.x;
      if (i < n) {
        int64_t tm = ts[i] + ts_base;
        if (tm > local_max) local_max = tm;
        int64_t win = (tm - align_ms) / len_ms;
        uint64_t packed =
            ((uint64_t)(uint32_t)(int32_t)win << 32) | (uint32_t)keys[i];
        int cb = (int)(region_of(mix64(packed), mask, region_bits) >> shift);
        int r = atomicAdd(&lcnt[cb], 1);
        if (r < V2_STAGE) {
          stage[cb][r] = packed;
        } else {
          // Tile-local stage overflow: straight to the residual area.
          int rp = atomicAdd(&gres[cb], 1);
          if (rp < res_cap) {
            ev_res[(int64_t)cb * res_cap + rp] = packed;
          } else {
            atomicExch(error_flag, 1);
          }
        }
      }
    }
    __syncthreads();
    // Flush complete 8-groups; keep the remainder staged.
    for (int b = threadIdx.x; b < V2_COARSE; b += blockDim.x) {
      int c = lcnt[b];
      if (c > V2_STAGE) c = V2_STAGE;
      int groups = c / 8;
      for (int g = 0; g < groups; ++g) {
        int base8 = atomicAdd(&gcur[b], 8);
        if (base8 + 8 <= cap_c) {
          uint64_t* dst = ev + (int64_t)b * cap_c + base8;
          #pragma unroll
          for (int q = 0; q < 8; ++q) dst[q] = stage[b][g * 8 + q];
        } else {
          for (int q = 0; q < 8; ++q) {
            int rp = atomicAdd(&gres[b], 1);
            if (rp < res_cap) {
              ev_res[(int64_t)b * res_cap + rp] = stage[b][g * 8 + q];
            } else {
              atomicExch(error_flag, 1);
            }
          }
        }
      }
      int rem = c - groups * 8;
      for (int q = 0; q < rem; ++q) {
        stage[b][q] = stage[b][groups * 8 + q];
      }
      lcnt[b] = rem;
    }
    __syncthreads();
  }
  // Final residual flush.
  for (int b = threadIdx.x; b < V2_COARSE; b += blockDim.x) {
    int c = lcnt[b];
    if (c > V2_STAGE) c = V2_STAGE;
    for (int q = 0; q < c; ++q) {
      int rp = atomicAdd(&gres[b], 1);
      if (rp < res_cap) {
        ev_res[(int64_t)b * res_cap + rp] = stage[b][q];
      } else {
        atomicExch(error_flag, 1);
      }
    }
  }
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    int64_t other = __shfl_down((long long)local_max, off);
    if (other > local_max) local_max = other;
  }
  if ((threadIdx.x & (WAVE - 1)) == 0 && local_max > 0) {
    atomicMax(max_ts, (unsigned long long)local_max);
  }
}

#define V2_CACHE 4096  // LDS cache entries in pass B

__global__ __launch_bounds__(256) void k_agg_cached(
    const uint64_t* __restrict__ ev,
    const uint64_t* __restrict__ ev_res,
    const int* __restrict__ gcur,
    const int* __restrict__ gres,
    int64_t cap_c,
    int64_t res_cap,
    int slices,  // blocks per coarse bucket
    uint64_t* __restrict__ tkeys,
    unsigned long long* __restrict__ tvals,
    uint64_t mask,
    int region_bits,
    int* __restrict__ error_flag) {
  __shared__ uint64_t ckeys[V2_CACHE];
  __shared__ unsigned long long cvals[V2_CACHE];
  for (int s = threadIdx.x; s < V2_CACHE; s += blockDim.x) {
    ckeys[s] = EMPTY_SLOT;
    cvals[s] = 0;
  }
  __syncthreads();
  int cb = blockIdx.x % V2_COARSE;
  int slice = blockIdx.x / V2_COARSE;

  auto agg_one = [&](uint64_t packed) {
    uint64_t h64 = mix64(packed);
    int lh = (int)((h64 >> 20) & (V2_CACHE - 1));
    // Two-probe LDS cache; collisions go straight to the table.
    for (int p = 0; p < 2; ++p) {
      uint64_t cur = ckeys[lh];
      if (cur == packed) {
        atomicAdd(&cvals[lh], 1ULL);
        return;
      }
      if (cur == EMPTY_SLOT) {
        uint64_t prev = atomicCAS(
            (unsigned long long*)&ckeys[lh], EMPTY_SLOT, packed);
        if (prev == EMPTY_SLOT || prev == packed) {
          atomicAdd(&cvals[lh], 1ULL);
          return;
        }
      }
      lh = (lh + 1) & (V2_CACHE - 1);
    }
    if (!hash_add(tkeys, tvals, mask, region_bits, packed, 1ULL)) {
      atomicExch(error_flag, 1);
    }
  };

  int c = gcur[cb];
  if (c > cap_c) c = (int)cap_c;
  for (int j = slice * (int)blockDim.x + threadIdx.x; j < c;
       j += slices * (int)blockDim.x) {
    agg_one(ev[(int64_t)cb * cap_c + j]);
  }
  int cr = gres[cb];
  if (cr > res_cap) cr = (int)res_cap;
  for (int j = slice * (int)blockDim.x + threadIdx.x; j < cr;
       j += slices * (int)blockDim.x) {
    agg_one(ev_res[(int64_t)cb * res_cap + j]);
  }
  __syncthreads();
  for (int s = threadIdx.x; s < V2_CACHE; s += blockDim.x) {
    if (ckeys[s] != EMPTY_SLOT) {
      if (!hash_add(tkeys, tvals, mask, region_bits, ckeys[s], cvals[s])) {
        atomicExch(error_flag, 1);
      }
    }
  }
}

// One-pass variant: events scatter into fixed-capacity per-region
// buffers (capacity `cap` each, laid out at region*cap), which removes
// the separate counting pass and the offsets scan.  A block's chunk
// that would overflow its region's buffer goes to the overflow spill
// (aggregated by k_radix_agg's caller via the contended-direct path —
// statistically empty for uniform keys at cap = 2x mean).
// GCN has no integer divide; `x / len_ms` compiles to a ~60-cycle
// sequence and the scatter does it twice per event.  The host
// precomputes a Granlund-Montgomery magic multiplier instead:
// win = mulhi64(x, m2) is exact for 0 <= x < maxfast (m2 == 0 or
// x outside the window falls back to the hardware-free division).
// Oldest window overlapping t for a sliding windower: the window ids
// are strides of `off`; an event at t lands in ids
// [floor((t-align-len)/off)+1, (t-align)/off].
__device__ __forceinline__ int64_t win_lo_of(
    int64_t t, int64_t align_ms, int64_t len_ms, int64_t off_ms) {
  int64_t num = t - align_ms - len_ms;
  int64_t fl = num >= 0 ? num / off_ms : -((-num + off_ms - 1) / off_ms);
  return fl + 1;
}

__device__ __forceinline__ int64_t win_of(
    int64_t t, int64_t align_ms, int64_t len_ms, uint64_t m2,
    uint64_t maxfast) {
  int64_t x = t - align_ms;
  if (m2 != 0 && (uint64_t)x < maxfast) {
    return (int64_t)__umul64hi((uint64_t)x, m2);
  }
  return x / len_ms;  // same truncation semantics as before
}

// Host side of the magic divider: p = 60 covers numerators < 2^60/e
// (ms-scale timestamps are < 2^43).  Power-of-two lengths reduce to a
// plain mulhi shift; len <= 1 disables the fast path.
static inline void magic_div_u64(
    int64_t d, uint64_t* m2, uint64_t* maxfast) {
  *m2 = 0;
  *maxfast = 0;
  if (d <= 1) return;
  if ((d & (d - 1)) == 0) {
    int s = 0;
    while ((int64_t(1) << s) < d) ++s;
    *m2 = (uint64_t)1 << (64 - s);
    *maxfast = ~(uint64_t)0;
    return;
  }
  const unsigned p = 60;
  unsigned __int128 one = 1;
  uint64_t M = (uint64_t)(((one << p) + (uint64_t)d - 1) / (uint64_t)d);
  uint64_t e = (uint64_t)((unsigned __int128)M * (uint64_t)d - (one << p));
  *m2 = M << 4;  // M < 2^60 for d >= 2, so the shift is exact
  *maxfast = e ? (uint64_t)((one << p) / e) : ~(uint64_t)0;
}

// TS is int64_t (absolute ms) or int32_t (deltas from `ts_base`, the
// wire format of the RCCL exchange — inserting the received segments
// directly skips the int64 timestamp rebuild entirely).
template <int MODE, typename TS = int64_t>
__global__ void k_radix_scatter_fixed(
    const int32_t* __restrict__ keys,
    const TS* __restrict__ ts,
    const int64_t* __restrict__ vals,
    int64_t n,
    int64_t align_ms,
    int64_t len_ms,
    int64_t off_ms,  // sliding stride; == len_ms for tumbling
    int64_t ts_base,
    uint64_t mask,
    int region_bits,
    int64_t cap,
    int* __restrict__ gcursors,       // [n_regions], pre-zeroed
    uint64_t* __restrict__ ev_packed,  // [n_regions * cap]
    int64_t* __restrict__ ev_vals,
    int* __restrict__ ov_cursor,       // [1], pre-zeroed
    uint64_t* __restrict__ ov_packed,  // overflow spill
    int64_t* __restrict__ ov_vals,
    int64_t ov_cap,
    unsigned long long* __restrict__ max_ts,
    int* __restrict__ error_flag,
    uint64_t win_m2,
    uint64_t win_maxfast) {
  extern __shared__ int lmem[];
  int nb = (int)(((mask + 1) >> region_bits));
  int* lhist = lmem;
  int* lbase = lmem + nb;
  for (int b = threadIdx.x; b < nb; b += blockDim.x) lhist[b] = 0;
  __syncthreads();
  int64_t start = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  int64_t local_max = 0;
  for (int64_t i = start; i < n; i += stride) {
    int64_t t = (int64_t)ts[i] + ts_base;
    if (t > local_max) local_max = t;
    uint64_t packed;
    int nwin = 1;
    if (MODE == AGG_TS) {
      packed = (uint64_t)(uint32_t)keys[i];
    } else {
      int64_t win = win_of(t, align_ms, off_ms, win_m2, win_maxfast);
      packed = ((uint64_t)(uint32_t)(int32_t)win << 32) | (uint32_t)keys[i];
      if (off_ms < len_ms) {
        nwin = (int)(win - win_lo_of(t, align_ms, len_ms, off_ms)) + 1;
      }
    }
    for (int w = 0; w < nwin; ++w) {
      atomicAdd(&lhist[(int)region_of(mix64(packed), mask, region_bits)], 1);
      packed -= (uint64_t)1 << 32;
    }
  }
  __syncthreads();
  for (int b = threadIdx.x; b < nb; b += blockDim.x) {
    // Unbounded reservation (one atomic per block x bucket; a bounded
    // CAS loop here storms under 2048-way contention).  Items whose
    // in-bucket position lands past `cap` spill to overflow; a block
    // still fills [base, cap) contiguously so the buffer has no holes
    // and readers clamp the cursor to `cap`.
    lbase[b] = lhist[b] > 0 ? atomicAdd(&gcursors[b], lhist[b]) : 0;
    lhist[b] = 0;
  }
  __syncthreads();
  for (int64_t i = start; i < n; i += stride) {
    int64_t t = (int64_t)ts[i] + ts_base;
    uint64_t packed;
    int nwin = 1;
    if (MODE == AGG_TS) {
      packed = (uint64_t)(uint32_t)keys[i];
    } else {
      int64_t win = win_of(t, align_ms, off_ms, win_m2, win_maxfast);
      packed = ((uint64_t)(uint32_t)(int32_t)win << 32) | (uint32_t)keys[i];
      if (off_ms < len_ms) {
        nwin = (int)(win - win_lo_of(t, align_ms, len_ms, off_ms)) + 1;
      }
    }
    int64_t v = (MODE == AGG_SUM) ? vals[i] : (MODE == AGG_TS ? t : 0);
    for (int w = 0; w < nwin; ++w) {
      int b = (int)region_of(mix64(packed), mask, region_bits);
      int64_t in_bucket = lbase[b] + atomicAdd(&lhist[b], 1);
      if (in_bucket < cap) {
        int64_t pos = (int64_t)b * cap + in_bucket;
        ev_packed[pos] = packed;
        if (MODE != AGG_COUNT) ev_vals[pos] = v;
      } else {
        int opos = atomicAdd(ov_cursor, 1);
        if (opos < ov_cap) {
          ov_packed[opos] = packed;
          if (MODE != AGG_COUNT) ov_vals[opos] = v;
        } else {
          atomicExch(error_flag, 1);
        }
      }
      packed -= (uint64_t)1 << 32;
    }
  }
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    int64_t other = __shfl_down((long long)local_max, off);
    if (other > local_max) local_max = other;
  }
  if ((threadIdx.x & (WAVE - 1)) == 0 && local_max > 0) {
    atomicMax(max_ts, (unsigned long long)local_max);
  }
}

// Direct one-pass scatter variant: one global atomicAdd on the
// region cursor per event, single read of the input (the block-hist
// version reads the input twice and does two LDS atomics per event).
// The region cursors are ~n_regions hot words in L2; with ≥2^11
// regions the per-address serialization is far below the memory
// time.  Selected via BYTEWAX_SCATTER_DIRECT (measured A/B; see
// profiles/).
template <int MODE, typename TS = int64_t>
__global__ void k_radix_scatter_direct(
    const int32_t* __restrict__ keys,
    const TS* __restrict__ ts,
    const int64_t* __restrict__ vals,
    int64_t n,
    int64_t align_ms,
    int64_t len_ms,
    int64_t off_ms,  // accepted for launch-shape parity; the launcher
                     // routes sliding (off < len) to the fixed variant
    int64_t ts_base,
    uint64_t mask,
    int region_bits,
    int64_t cap,
    int* __restrict__ gcursors,
    uint64_t* __restrict__ ev_packed,
    int64_t* __restrict__ ev_vals,
    int* __restrict__ ov_cursor,
    uint64_t* __restrict__ ov_packed,
    int64_t* __restrict__ ov_vals,
    int64_t ov_cap,
    unsigned long long* __restrict__ max_ts,
    int* __restrict__ error_flag,
    uint64_t win_m2,
    uint64_t win_maxfast) {
  int64_t start = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  int64_t local_max = 0;
  for (int64_t i = start; i < n; i += stride) {
    int64_t t = (int64_t)ts[i] + ts_base;
    if (t > local_max) local_max = t;
    int64_t win = win_of(t, align_ms, len_ms, win_m2, win_maxfast);
    uint64_t packed =
        ((uint64_t)(uint32_t)(int32_t)win << 32) | (uint32_t)keys[i];
    int b = (int)region_of(mix64(packed), mask, region_bits);
    int64_t in_bucket = atomicAdd(&gcursors[b], 1);
    int64_t v = (MODE == AGG_SUM) ? vals[i] : 0;
    if (in_bucket < cap) {
      int64_t pos = (int64_t)b * cap + in_bucket;
      ev_packed[pos] = packed;
      if (MODE == AGG_SUM) ev_vals[pos] = v;
    } else {
      int opos = atomicAdd(ov_cursor, 1);
      if (opos < ov_cap) {
        ov_packed[opos] = packed;
        if (MODE == AGG_SUM) ov_vals[opos] = v;
      } else {
        atomicExch(error_flag, 1);
      }
    }
  }
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    int64_t other = __shfl_down((long long)local_max, off);
    if (other > local_max) local_max = other;
  }
  if ((threadIdx.x & (WAVE - 1)) == 0 && local_max > 0) {
    atomicMax(max_ts, (unsigned long long)local_max);
  }
}

// Line-staged scatter: kills the partial-line write amplification of
// the scattered 8 B segment writes (round-1 PMC: 7.7x — ~2 GB of
// actual HBM writes for a 256 MB payload).  Two invariants make every
// `ev_packed` line a single full-line HBM write:
//   1. the shared segment cursors advance only in SC_GRAN-event
//      (64 B) units, so each line is exclusively owned by the one
//      block that reserved it (no cross-XCD line sharing);
//   2. a block writes all 8 entries of its granule within one tile
//      phase, so the line completes in the local XCD's L2 (and is
//      evicted once, whole) instead of staying open across the whole
//      input sweep.
// Events stage per segment in LDS; residual (< SC_GRAN) events per
// segment ride along across tiles and flush as sentinel-padded
// granules at the end (the aggregation kernels skip EMPTY_SLOT).
// Segments may be COARSER than table regions (seg_bits >
// region_bits): fewer, longer runs; the agg kernel then covers
// 2^(seg_bits-region_bits) regions per block.
#define SC_GRAN 8  // events per cursor reservation = one 64 B line

template <int MODE, typename TS = int64_t, int U = 16, int BLK = 256>
__global__ __launch_bounds__(BLK) void k_radix_scatter_staged(
    const int32_t* __restrict__ keys,
    const TS* __restrict__ ts,
    const int64_t* __restrict__ vals,  // nullptr for COUNT
    int64_t n,
    int64_t align_ms,
    int64_t len_ms,
    int64_t off_ms,  // sliding stride; == len_ms for tumbling
    int64_t ts_base,
    uint64_t mask,
    int seg_bits,
    int64_t cap,  // per-segment capacity; multiple of SC_GRAN
    int* __restrict__ gcursors,        // [nseg], zeroed before 1st call
    uint64_t* __restrict__ ev_packed,  // [nseg * cap]
    int64_t* __restrict__ ev_vals,     // [nseg * cap] (SUM)
    int* __restrict__ ov_cursor,
    uint64_t* __restrict__ ov_packed,
    int64_t* __restrict__ ov_vals,
    int64_t ov_cap,
    unsigned long long* __restrict__ max_ts,
    int* __restrict__ error_flag,
    uint64_t win_m2,
    uint64_t win_maxfast) {
  extern __shared__ char smem[];
  const int nseg = (int)(((mask + 1) >> seg_bits));
  uint64_t* res = (uint64_t*)smem;  // [nseg][SC_GRAN] residual staging
  int64_t* res_v =
      (int64_t*)(smem + (size_t)nseg * SC_GRAN * 8);  // SUM/TS only
  int* res_cnt =
      (int*)(smem + (size_t)nseg * SC_GRAN * 8 *
                        (MODE != AGG_COUNT ? 2 : 1));  // [nseg]
  int* lhist = res_cnt + nseg;  // [nseg] tile hist, then grant
  int* lbase = lhist + nseg;    // [nseg] granted global base
  int* lofs = lbase + nseg;     // [nseg] virtual-position cursor
  for (int b = threadIdx.x; b < nseg; b += blockDim.x) {
    res_cnt[b] = 0;
    lhist[b] = 0;
  }
  __syncthreads();

  // One granule write, with capacity split to the overflow spill.
  auto emit = [&](int b, int64_t gpos, uint64_t packed, int64_t v) {
    if (gpos < cap) {
      int64_t pos = (int64_t)b * cap + gpos;
      ev_packed[pos] = packed;
      if (MODE != AGG_COUNT) ev_vals[pos] = v;
    } else if (packed != EMPTY_SLOT) {
      int opos = atomicAdd(ov_cursor, 1);
      if (opos < ov_cap) {
        ov_packed[opos] = packed;
        if (MODE != AGG_COUNT) ov_vals[opos] = v;
      } else {
        atomicExch(error_flag, 1);
      }
    }
  };

  const int64_t tile = (int64_t)blockDim.x * U;
  int64_t local_max = 0;
  for (int64_t t0 = (int64_t)blockIdx.x * tile; t0 < n;
       t0 += (int64_t)gridDim.x * tile) {
    uint64_t pk[U];
    int64_t pv[U];
    int sg[U];
    int nw[U];
    // P1: load, classify, histogram.
    for (int u = 0; u < U; ++u) {
      int64_t i = t0 + (int64_t)u * blockDim.x + threadIdx.x;
      sg[u] = -1;
      if (i >= n) continue;
      int64_t t = (int64_t)ts[i] + ts_base;
      if (t > local_max) local_max = t;
      uint64_t packed;
      int nwin = 1;
      if (MODE == AGG_TS) {
        packed = (uint64_t)(uint32_t)keys[i];
      } else {
        // `win_m2` is the magic divisor of the STRIDE (off_ms); for a
        // sliding windower the event expands into every window id in
        // [win_lo, win].
        int64_t win = win_of(t, align_ms, off_ms, win_m2, win_maxfast);
        packed =
            ((uint64_t)(uint32_t)(int32_t)win << 32) | (uint32_t)keys[i];
        if (off_ms < len_ms) {
          nwin =
              (int)(win - win_lo_of(t, align_ms, len_ms, off_ms)) + 1;
        }
      }
      pk[u] = packed;
      nw[u] = nwin;
      if (MODE == AGG_SUM) pv[u] = vals[i];
      else if (MODE == AGG_TS) pv[u] = t;
      int b = (int)region_of(mix64(packed), mask, seg_bits);
      sg[u] = b;
      atomicAdd(&lhist[b], 1);
      // Expanded entries get their own histogram slots.
      uint64_t p2 = packed;
      for (int w = 1; w < nwin; ++w) {
        p2 -= (uint64_t)1 << 32;
        atomicAdd(
            &lhist[(int)region_of(mix64(p2), mask, seg_bits)], 1);
      }
    }
    __syncthreads();
    // P2: per-segment granule reservation.
    for (int b = threadIdx.x; b < nseg; b += blockDim.x) {
      int tot = res_cnt[b] + lhist[b];
      int grant = tot & ~(SC_GRAN - 1);
      lbase[b] = grant > 0 ? atomicAdd(&gcursors[b], grant) : 0;
      lofs[b] = res_cnt[b];
      lhist[b] = grant;
    }
    __syncthreads();
    // P3a: drain old residual into the granted granule (granule >=
    // SC_GRAN > residual when granted, so all-or-nothing).
    for (int b = threadIdx.x; b < nseg; b += blockDim.x) {
      int grant = lhist[b];
      if (grant > 0) {
        int rc = res_cnt[b];
        int gb = lbase[b];
        for (int j = 0; j < rc; ++j) {
          emit(b, (int64_t)gb + j, res[(size_t)b * SC_GRAN + j],
               MODE != AGG_COUNT ? res_v[(size_t)b * SC_GRAN + j] : 0);
        }
        res_cnt[b] = 0;
      }
    }
    __syncthreads();
    // P3b: place this tile's (possibly window-expanded) entries at
    // their virtual positions.
    for (int u = 0; u < U; ++u) {
      if (sg[u] < 0) continue;
      uint64_t p2 = pk[u];
      for (int w = 0; w < nw[u]; ++w) {
        int b = w == 0 ? sg[u]
                       : (int)region_of(mix64(p2), mask, seg_bits);
        int vpos = atomicAdd(&lofs[b], 1);
        int grant = lhist[b];
        if (vpos < grant) {
          emit(b, (int64_t)lbase[b] + vpos, p2,
               MODE != AGG_COUNT ? pv[u] : 0);
        } else {
          res[(size_t)b * SC_GRAN + (vpos - grant)] = p2;
          if (MODE != AGG_COUNT)
            res_v[(size_t)b * SC_GRAN + (vpos - grant)] = pv[u];
        }
        p2 -= (uint64_t)1 << 32;
      }
    }
    __syncthreads();
    // P3c: carry the leftover count; reset the histogram.
    for (int b = threadIdx.x; b < nseg; b += blockDim.x) {
      res_cnt[b] = lofs[b] - lhist[b];
      lhist[b] = 0;
    }
    __syncthreads();
  }
  // Final flush: pad each non-empty residual to a full granule with
  // EMPTY_SLOT sentinels (skipped by the agg kernels).
  for (int b = threadIdx.x; b < nseg; b += blockDim.x) {
    int rc = res_cnt[b];
    if (rc == 0) continue;
    int gb = atomicAdd(&gcursors[b], SC_GRAN);
    for (int j = 0; j < SC_GRAN; ++j) {
      uint64_t p = j < rc ? res[(size_t)b * SC_GRAN + j] : EMPTY_SLOT;
      emit(b, (int64_t)gb + j, p,
           (MODE != AGG_COUNT && j < rc) ? res_v[(size_t)b * SC_GRAN + j]
                                         : 0);
    }
  }
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    int64_t other = __shfl_down((long long)local_max, off);
    if (other > local_max) local_max = other;
  }
  if ((threadIdx.x & (WAVE - 1)) == 0 && local_max > 0) {
    atomicMax(max_ts, (unsigned long long)local_max);
  }
}


// Coalesced-placement variant of the staged scatter (COUNT,
// tumbling): stage the tile's granted entries in LDS together with
// their precomputed global positions, then write them back with a
// LINEAR copy loop — consecutive lanes hit consecutive addresses
// inside each segment's granule run, so the hardware coalesces ~8
// scattered-by-segment events into one full-line transaction instead
// of 8.  Probes the hypothesis that the scatter is bound by L2
// random-store transaction rate (~1 tx/event in the default staged
// kernel), not bytes.  Selected via BYTEWAX_SCATTER=staged2.
template <typename TS = int64_t, int U = 16>
__global__ __launch_bounds__(256) void k_radix_scatter_staged2(
    const int32_t* __restrict__ keys,
    const TS* __restrict__ ts,
    const int64_t* __restrict__ vals,  // unused (COUNT)
    int64_t n,
    int64_t align_ms,
    int64_t len_ms,
    int64_t off_ms,
    int64_t ts_base,
    uint64_t mask,
    int seg_bits,
    int64_t cap,
    int* __restrict__ gcursors,
    uint64_t* __restrict__ ev_packed,
    int64_t* __restrict__ ev_vals,  // unused (COUNT)
    int* __restrict__ ov_cursor,
    uint64_t* __restrict__ ov_packed,
    int64_t* __restrict__ ov_vals,  // unused (COUNT)
    int64_t ov_cap,
    unsigned long long* __restrict__ max_ts,
    int* __restrict__ error_flag,
    uint64_t win_m2,
    uint64_t win_maxfast) {
  extern __shared__ char smem[];
  const int nseg = (int)(((mask + 1) >> seg_bits));
  const int T = (int)blockDim.x * U;
  const int MAXSTAGE = T + 7 * 2048;  // upper bound; sized by host LDS
  uint64_t* stage = (uint64_t*)smem;                       // [maxstage]
  int* gpos = (int*)(smem + (size_t)(T + 7 * nseg) * 8);   // [maxstage]
  uint64_t* res = (uint64_t*)(smem + (size_t)(T + 7 * nseg) * 12);
  int* res_cnt = (int*)((char*)res + (size_t)nseg * SC_GRAN * 8);
  int* lhist = res_cnt + nseg;
  int* lbase = lhist + nseg;
  int* ltoff = lbase + nseg;
  int* lvirt = ltoff + nseg;
  int* lgrant = lvirt + nseg;
  int* ltotal = lgrant + nseg;  // [1]
  (void)MAXSTAGE;
  for (int b = threadIdx.x; b < nseg; b += blockDim.x) {
    res_cnt[b] = 0;
    lhist[b] = 0;
  }
  __syncthreads();

  auto spill = [&](uint64_t packed) {
    int opos = atomicAdd(ov_cursor, 1);
    if (opos < ov_cap) ov_packed[opos] = packed;
    else atomicExch(error_flag, 1);
  };

  const int64_t tile = (int64_t)blockDim.x * U;
  int64_t local_max = 0;
  for (int64_t t0 = (int64_t)blockIdx.x * tile; t0 < n;
       t0 += (int64_t)gridDim.x * tile) {
    uint64_t pk[U];
    int sg[U];
    // P1: classify + histogram.
    for (int u = 0; u < U; ++u) {
      int64_t i = t0 + (int64_t)u * blockDim.x + threadIdx.x;
      sg[u] = -1;
      if (i >= n) continue;
      int64_t t = (int64_t)ts[i] + ts_base;
      if (t > local_max) local_max = t;
      int64_t win = win_of(t, align_ms, len_ms, win_m2, win_maxfast);
      uint64_t packed =
          ((uint64_t)(uint32_t)(int32_t)win << 32) | (uint32_t)keys[i];
      pk[u] = packed;
      int b = (int)region_of(mix64(packed), mask, seg_bits);
      sg[u] = b;
      atomicAdd(&lhist[b], 1);
    }
    __syncthreads();
    // P2: granule grants + in-tile staging offsets (exclusive scan of
    // the per-segment grants).
    for (int b = threadIdx.x; b < nseg; b += blockDim.x) {
      int tot = res_cnt[b] + lhist[b];
      int grant = tot & ~(SC_GRAN - 1);
      lgrant[b] = grant;
      lbase[b] = grant > 0 ? atomicAdd(&gcursors[b], grant) : 0;
      // Virtual positions count residual + new; ungranted segments
      // keep their residual in place and append after it.
      lvirt[b] = res_cnt[b];
      ltoff[b] = grant;  // scanned in place below
    }
    __syncthreads();
    // Hillis-Steele inclusive scan over ltoff[nseg], then shift to
    // exclusive via (incl - grant).
    for (int d = 1; d < nseg; d <<= 1) {
      int v[8];
      int nb = 0;
      for (int b = threadIdx.x; b < nseg; b += blockDim.x) {
        v[nb++] = b >= d ? ltoff[b - d] : 0;
      }
      __syncthreads();
      nb = 0;
      for (int b = threadIdx.x; b < nseg; b += blockDim.x) {
        ltoff[b] += v[nb++];
      }
      __syncthreads();
    }
    if (threadIdx.x == 0) ltotal[0] = ltoff[nseg - 1];
    __syncthreads();
    // P3a: drain old residual into staging (virtual slots 0..rc).
    for (int b = threadIdx.x; b < nseg; b += blockDim.x) {
      int grant = lgrant[b];
      if (grant > 0) {
        int rc = res_cnt[b];
        int off = ltoff[b] - grant;
        int gb = lbase[b];
        for (int j = 0; j < rc; ++j) {
          int64_t gp = (int64_t)gb + j;
          if (gp < cap) {
            stage[off + j] = res[(size_t)b * SC_GRAN + j];
            gpos[off + j] = (int)((int64_t)b * cap + gp);
          } else {
            stage[off + j] = 0;
            gpos[off + j] = -1;
            spill(res[(size_t)b * SC_GRAN + j]);
          }
        }
      }
    }
    __syncthreads();
    // P3b: place new events (staged if granted, else residual).
    for (int u = 0; u < U; ++u) {
      int b = sg[u];
      if (b < 0) continue;
      int virt = atomicAdd(&lvirt[b], 1);
      int grant = lgrant[b];
      if (virt < grant) {
        int off = ltoff[b] - grant + virt;
        int64_t gp = (int64_t)lbase[b] + virt;
        if (gp < cap) {
          stage[off] = pk[u];
          gpos[off] = (int)((int64_t)b * cap + gp);
        } else {
          stage[off] = 0;
          gpos[off] = -1;
          spill(pk[u]);
        }
      } else {
        res[(size_t)b * SC_GRAN + (virt - grant)] = pk[u];
      }
    }
    __syncthreads();
    // P4: coalesced copy-out — consecutive threads write consecutive
    // staged slots; slots are segment-grouped and granule-aligned, so
    // adjacent lanes hit adjacent addresses of the same 64 B lines.
    int total = ltotal[0];
    for (int j = threadIdx.x; j < total; j += blockDim.x) {
      int gp = gpos[j];
      if (gp >= 0) ev_packed[gp] = stage[j];
    }
    // P5: carry leftovers.
    for (int b = threadIdx.x; b < nseg; b += blockDim.x) {
      res_cnt[b] = lvirt[b] - lgrant[b];
      if (res_cnt[b] < 0) res_cnt[b] = 0;
      lhist[b] = 0;
    }
    __syncthreads();
  }
  // Final flush: sentinel-padded granules, like the default staged.
  for (int b = threadIdx.x; b < nseg; b += blockDim.x) {
    int rc = res_cnt[b];
    if (rc == 0) continue;
    int gb = atomicAdd(&gcursors[b], SC_GRAN);
    for (int j = 0; j < SC_GRAN; ++j) {
      uint64_t p = j < rc ? res[(size_t)b * SC_GRAN + j] : EMPTY_SLOT;
      int64_t gp = (int64_t)gb + j;
      if (gp < cap) ev_packed[(int64_t)b * cap + gp] = p;
      else if (p != EMPTY_SLOT) spill(p);
    }
  }
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    int64_t other = __shfl_down((long long)local_max, off);
    if (other > local_max) local_max = other;
  }
  if ((threadIdx.x & (WAVE - 1)) == 0 && local_max > 0) {
    atomicMax(max_ts, (unsigned long long)local_max);
  }
}

// Aggregate the overflow spill straight into the table (tiny for
// uniform keys).
template <int MODE>
__global__ void k_overflow_agg(
    const uint64_t* __restrict__ ov_packed,
    const int64_t* __restrict__ ov_vals,
    const int* __restrict__ ov_cursor,
    int64_t ov_cap,
    uint64_t* __restrict__ tkeys,
    unsigned long long* __restrict__ tvals,
    uint64_t mask,
    int region_bits,
    int* __restrict__ error_flag) {
  int64_t n = *ov_cursor;
  if (n > ov_cap) n = ov_cap;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    unsigned long long inc =
        (MODE == AGG_SUM) ? (unsigned long long)ov_vals[i] : 1ULL;
    if (!hash_add(tkeys, tvals, mask, region_bits, ov_packed[i], inc)) {
      atomicExch(error_flag, 1);
    }
  }
}

template <int MODE>
__global__ __launch_bounds__(1024) void k_radix_agg(
    const uint64_t* __restrict__ ev_packed,
    const int64_t* __restrict__ ev_vals,
    const int* __restrict__ offsets,
    const int* __restrict__ counts,
    int64_t clamp_cap,  // 0 = counts are exact; >0 = clamp (fixed layout)
    uint64_t* __restrict__ tkeys,
    unsigned long long* __restrict__ tvals,
    uint64_t mask,
    int region_bits,
    int lds_bits,  // LDS staging-table slots (>= region_bits when one
                   // scatter segment spans several table regions)
    int* __restrict__ error_flag) {
  extern __shared__ char smem[];
  int region = 1 << lds_bits;
  uint64_t* lkeys = (uint64_t*)smem;
  unsigned long long* lvals =
      (unsigned long long*)(smem + (size_t)region * sizeof(uint64_t));
  for (int s = threadIdx.x; s < region; s += blockDim.x) {
    lkeys[s] = EMPTY_SLOT;
    lvals[s] = 0;
  }
  __syncthreads();
  int b = blockIdx.x;
  int cnt = counts[b];
  if (clamp_cap > 0 && cnt > (int)clamp_cap) cnt = (int)clamp_cap;
  int start = offsets[b];
  for (int j = threadIdx.x; j < cnt; j += blockDim.x) {
    uint64_t packed = ev_packed[start + j];
    if (packed == EMPTY_SLOT) continue;  // staged-scatter pad
    unsigned long long inc =
        (MODE == AGG_SUM) ? (unsigned long long)ev_vals[start + j] : 1ULL;
    uint64_t h64 = mix64(packed);
    int lh = (int)((h64 >> 32) & (region - 1));
    bool done = false;
    for (int p = 0; p < region; ++p) {
      uint64_t cur = lkeys[lh];
      if (cur == packed) {
        atomicAdd(&lvals[lh], inc);
        done = true;
        break;
      }
      if (cur == EMPTY_SLOT) {
        uint64_t prev = atomicCAS(
            (unsigned long long*)&lkeys[lh], EMPTY_SLOT, packed);
        if (prev == EMPTY_SLOT || prev == packed) {
          atomicAdd(&lvals[lh], inc);
          done = true;
          break;
        }
      }
      lh = (lh + 1) & (region - 1);
    }
    if (!done) {
      // LDS staging full (only possible when the global regions are
      // at least as full): flush straight to the global region.
      if (!hash_add(tkeys, tvals, mask, region_bits, packed, inc)) {
        atomicExch(error_flag, 1);
      }
    }
  }
  __syncthreads();
  // Flush distinct keys once into each key's own table region (the
  // segment's regions are contiguous in HBM, so flushes stay local).
  for (int s = threadIdx.x; s < region; s += blockDim.x) {
    if (lkeys[s] != EMPTY_SLOT) {
      if (!hash_add(tkeys, tvals, mask, region_bits, lkeys[s], lvals[s])) {
        atomicExch(error_flag, 1);
      }
    }
  }
}

// Overflow spill for the stats table (direct contended path).
__global__ void k_overflow_agg_stats(
    const uint64_t* __restrict__ ov_packed,
    const int64_t* __restrict__ ov_vals,
    const int* __restrict__ ov_cursor,
    int64_t ov_cap,
    uint64_t* __restrict__ tkeys,
    long long* __restrict__ tcnt,
    long long* __restrict__ tsum,
    long long* __restrict__ tmin,
    long long* __restrict__ tmax,
    uint64_t mask,
    int* __restrict__ error_flag) {
  int64_t n = *ov_cursor;
  if (n > ov_cap) n = ov_cap;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    uint64_t slot = find_slot(tkeys, mask, ov_packed[i]);
    if (slot == ~0ULL) {
      atomicExch(error_flag, 1);
      continue;
    }
    long long v = ov_vals[i];
    atomicAdd((unsigned long long*)&tcnt[slot], 1ULL);
    atomicAdd((unsigned long long*)&tsum[slot], (unsigned long long)v);
    atomicMin(&tmin[slot], v);
    atomicMax(&tmax[slot], v);
  }
}

// Radix aggregation for the 4-accumulator stats table (count / sum /
// min / max): one workgroup per region, LDS-resident accumulators,
// one flush per distinct cell.  Events must already be partitioned
// (k_radix_scatter with values).
__global__ __launch_bounds__(1024) void k_radix_agg_stats(
    const uint64_t* __restrict__ ev_packed,
    const int64_t* __restrict__ ev_vals,
    const int* __restrict__ offsets,
    const int* __restrict__ counts,
    uint64_t* __restrict__ tkeys,
    long long* __restrict__ tcnt,
    long long* __restrict__ tsum,
    long long* __restrict__ tmin,
    long long* __restrict__ tmax,
    int64_t clamp_cap,
    uint64_t mask,
    int region_bits,
    int lds_bits,  // LDS staging slots (>= region_bits for coarse segments)
    int* __restrict__ error_flag) {
  extern __shared__ char smem[];
  int region = 1 << lds_bits;
  uint64_t* lkeys = (uint64_t*)smem;
  long long* lcnt = (long long*)(smem + (size_t)region * 8);
  long long* lsum = (long long*)(smem + (size_t)region * 16);
  long long* lmin = (long long*)(smem + (size_t)region * 24);
  long long* lmax = (long long*)(smem + (size_t)region * 32);
  const long long LLMAX = 0x7FFFFFFFFFFFFFFFLL;
  for (int s = threadIdx.x; s < region; s += blockDim.x) {
    lkeys[s] = EMPTY_SLOT;
    lcnt[s] = 0;
    lsum[s] = 0;
    lmin[s] = LLMAX;
    lmax[s] = -LLMAX - 1;
  }
  __syncthreads();
  int b = blockIdx.x;
  int cnt = counts[b];
  if (clamp_cap > 0 && cnt > (int)clamp_cap) cnt = (int)clamp_cap;
  int start = offsets[b];
  for (int j = threadIdx.x; j < cnt; j += blockDim.x) {
    uint64_t packed = ev_packed[start + j];
    if (packed == EMPTY_SLOT) continue;  // staged-scatter pad
    long long v = ev_vals[start + j];
    uint64_t h64 = mix64(packed);
    int lh = (int)((h64 >> 32) & (region - 1));
    int slot = -1;
    for (int p = 0; p < region; ++p) {
      uint64_t cur = lkeys[lh];
      if (cur == packed) {
        slot = lh;
        break;
      }
      if (cur == EMPTY_SLOT) {
        uint64_t prev = atomicCAS(
            (unsigned long long*)&lkeys[lh], EMPTY_SLOT, packed);
        if (prev == EMPTY_SLOT || prev == packed) {
          slot = lh;
          break;
        }
      }
      lh = (lh + 1) & (region - 1);
    }
    if (slot >= 0) {
      atomicAdd((unsigned long long*)&lcnt[slot], 1ULL);
      atomicAdd((unsigned long long*)&lsum[slot], (unsigned long long)v);
      atomicMin(&lmin[slot], v);
      atomicMax(&lmax[slot], v);
    } else {
      // LDS region full: fall through to the global region directly.
      uint64_t gslot = find_slot(tkeys, mask, packed);
      if (gslot == ~0ULL) {
        atomicExch(error_flag, 1);
      } else {
        atomicAdd((unsigned long long*)&tcnt[gslot], 1ULL);
        atomicAdd((unsigned long long*)&tsum[gslot], (unsigned long long)v);
        atomicMin(&tmin[gslot], v);
        atomicMax(&tmax[gslot], v);
      }
    }
  }
  __syncthreads();
  for (int s = threadIdx.x; s < region; s += blockDim.x) {
    if (lkeys[s] != EMPTY_SLOT) {
      uint64_t gslot = find_slot(tkeys, mask, lkeys[s]);
      if (gslot == ~0ULL) {
        atomicExch(error_flag, 1);
        continue;
      }
      atomicAdd((unsigned long long*)&tcnt[gslot],
                (unsigned long long)lcnt[s]);
      atomicAdd((unsigned long long*)&tsum[gslot],
                (unsigned long long)lsum[s]);
      atomicMin(&tmin[gslot], lmin[s]);
      atomicMax(&tmax[gslot], lmax[s]);
    }
  }
}

// Extract all slots whose window id is in [win_lo, win_hi) WITHOUT
// mutating the table.  Deleting slots in place is forbidden: it breaks
// open-addressing probe chains for keys displaced past the cleared
// slot (a later re-insert of such a key would claim its home slot and
// duplicate the cell).  Reclamation happens via k_close_migrate.
// Output compaction: wave ballot + one atomic per wave, lanes write at
// their popcount rank.
__global__ void k_close_extract(
    const uint64_t* __restrict__ tkeys,
    const unsigned long long* __restrict__ tvals,
    int64_t nslots,
    int64_t win_lo,
    int64_t win_hi,
    int32_t* __restrict__ out_keys,
    int32_t* __restrict__ out_wins,
    int64_t* __restrict__ out_vals,
    int* __restrict__ out_n,
    int64_t cap) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < nslots; i += stride) {
    uint64_t k = tkeys[i];
    bool take = false;
    int32_t win = 0;
    if (k != EMPTY_SLOT) {
      win = (int32_t)(uint32_t)(k >> 32);
      take = (int64_t)win >= win_lo && (int64_t)win < win_hi;
    }
    unsigned long long ball = __ballot(take);
    int lane = threadIdx.x & (WAVE - 1);
    if (ball != 0) {
      int wave_total = __popcll(ball);
      int rank = __popcll(ball & ((1ULL << lane) - 1ULL));
      int base = 0;
      int lead = __ffsll((unsigned long long)ball) - 1;
      if (lane == lead) base = atomicAdd(out_n, wave_total);
      base = __shfl(base, lead);
      if (take) {
        int64_t idx = base + rank;
        if (idx < cap) {
          out_keys[idx] = (int32_t)(uint32_t)(k & 0xFFFFFFFFULL);
          out_wins[idx] = win;
          out_vals[idx] = (int64_t)tvals[i];
        }
      }
    }
  }
}

// Window close with reclamation: emit cells below `win_hi` and
// migrate the still-open cells into a fresh (pre-reset) table.  The
// caller swaps tables afterwards and resets the old one
// asynchronously.  Migrated cells are unique by construction so the
// rebuild cannot create duplicate chains.
__global__ void k_close_migrate(
    const uint64_t* __restrict__ tkeys,
    const unsigned long long* __restrict__ tvals,
    int64_t nslots,
    int64_t win_hi,
    uint64_t* __restrict__ nkeys,
    unsigned long long* __restrict__ nvals,
    uint64_t mask,
    int region_bits,
    int32_t* __restrict__ out_keys,
    int32_t* __restrict__ out_wins,
    int64_t* __restrict__ out_vals,
    int* __restrict__ out_n,
    int64_t cap,
    int* __restrict__ error_flag) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < nslots; i += stride) {
    uint64_t k = tkeys[i];
    bool occupied = (k != EMPTY_SLOT);
    int32_t win = 0;
    bool take = false;
    if (occupied) {
      win = (int32_t)(uint32_t)(k >> 32);
      take = (int64_t)win < win_hi;
      if (!take) {
        if (!hash_add(nkeys, nvals, mask, region_bits, k, tvals[i])) {
          atomicExch(error_flag, 1);
        }
      }
    }
    unsigned long long ball = __ballot(take);
    int lane = threadIdx.x & (WAVE - 1);
    if (ball != 0) {
      int wave_total = __popcll(ball);
      int rank = __popcll(ball & ((1ULL << lane) - 1ULL));
      int base = 0;
      int lead = __ffsll((unsigned long long)ball) - 1;
      if (lane == lead) base = atomicAdd(out_n, wave_total);
      base = __shfl(base, lead);
      if (take) {
        int64_t idx = base + rank;
        if (idx < cap) {
          out_keys[idx] = (int32_t)(uint32_t)(k & 0xFFFFFFFFULL);
          out_wins[idx] = win;
          out_vals[idx] = (int64_t)tvals[i];
        }
      }
    }
  }
}

// Find (or claim) the slot for `packed` in an open-address table
// (whole-table or region layout, see hash_add).  Returns the slot
// index or ~0 on full table/region.
__device__ __forceinline__ uint64_t find_slot_r(
    uint64_t* __restrict__ tkeys, uint64_t mask, int region_bits,
    uint64_t packed) {
  uint64_t h64 = mix64(packed);
  uint64_t h, probe_mask, base;
  if (region_bits == 0) {
    base = 0;
    probe_mask = mask;
    h = h64 & mask;
  } else {
    base = region_of(h64, mask, region_bits) << region_bits;
    probe_mask = (1ULL << region_bits) - 1;
    h = (h64 >> 32) & probe_mask;
  }
  for (uint64_t probes = 0; probes <= probe_mask; ++probes) {
    uint64_t slot = base | h;
    uint64_t cur = tkeys[slot];
    if (cur == packed) return slot;
    if (cur == EMPTY_SLOT) {
      uint64_t prev = atomicCAS(
          (unsigned long long*)&tkeys[slot], EMPTY_SLOT, packed);
      if (prev == EMPTY_SLOT || prev == packed) return slot;
    }
    h = (h + 1) & probe_mask;
  }
  return ~0ULL;
}

__device__ __forceinline__ uint64_t find_slot(
    uint64_t* __restrict__ tkeys, uint64_t mask, uint64_t packed) {
  return find_slot_r(tkeys, mask, 0, packed);
}

// 1BRC-style keyed running stats: count / sum / min / max per
// (key, window).  One pass over the batch, four atomics per event.
__global__ void k_stats_insert(
    const int32_t* __restrict__ keys,
    const int64_t* __restrict__ ts,
    const int64_t* __restrict__ vals,
    int64_t n,
    uint64_t* __restrict__ tkeys,
    long long* __restrict__ tcnt,
    long long* __restrict__ tsum,
    long long* __restrict__ tmin,
    long long* __restrict__ tmax,
    uint64_t mask,
    int64_t align_ms,
    int64_t len_ms,
    int64_t ts_base,
    unsigned long long* __restrict__ max_ts,
    int* __restrict__ error_flag) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  int64_t local_max = 0;
  for (; i < n; i += stride) {
    int64_t t = ts[i] + ts_base;
    if (t > local_max) local_max = t;
    int64_t win = (t - align_ms) / len_ms;
    uint64_t packed =
        ((uint64_t)(uint32_t)(int32_t)win << 32) | (uint32_t)keys[i];
    uint64_t slot = find_slot(tkeys, mask, packed);
    if (slot == ~0ULL) {
      atomicExch(error_flag, 1);
      continue;
    }
    long long v = vals[i];
    atomicAdd((unsigned long long*)&tcnt[slot], 1ULL);
    atomicAdd((unsigned long long*)&tsum[slot], (unsigned long long)v);
    atomicMin(&tmin[slot], v);
    atomicMax(&tmax[slot], v);
  }
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    int64_t other = __shfl_down((long long)local_max, off);
    if (other > local_max) local_max = other;
  }
  if ((threadIdx.x & (WAVE - 1)) == 0 && local_max > 0) {
    atomicMax(max_ts, (unsigned long long)local_max);
  }
}

// Recovery fixup: add deltas to count/sum of existing slots.
__global__ void k_stats_fixup(
    const int32_t* __restrict__ keys,
    const int64_t* __restrict__ ts,
    const int64_t* __restrict__ cnt_delta,
    const int64_t* __restrict__ sum_delta,
    int64_t n,
    uint64_t* __restrict__ tkeys,
    long long* __restrict__ tcnt,
    long long* __restrict__ tsum,
    uint64_t mask,
    int64_t align_ms,
    int64_t len_ms) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    int64_t win = (ts[i] - align_ms) / len_ms;
    uint64_t packed =
        ((uint64_t)(uint32_t)(int32_t)win << 32) | (uint32_t)keys[i];
    uint64_t slot = find_slot(tkeys, mask, packed);
    if (slot == ~0ULL) continue;
    atomicAdd((unsigned long long*)&tcnt[slot],
              (unsigned long long)cnt_delta[i]);
    atomicAdd((unsigned long long*)&tsum[slot],
              (unsigned long long)sum_delta[i]);
  }
}

__global__ void k_stats_extract(
    const uint64_t* __restrict__ tkeys,
    const long long* __restrict__ tcnt,
    const long long* __restrict__ tsum,
    const long long* __restrict__ tmin,
    const long long* __restrict__ tmax,
    int64_t nslots,
    int64_t win_lo,
    int64_t win_hi,
    int32_t* __restrict__ out_keys,
    int32_t* __restrict__ out_wins,
    int64_t* __restrict__ out_cnt,
    int64_t* __restrict__ out_sum,
    int64_t* __restrict__ out_min,
    int64_t* __restrict__ out_max,
    int* __restrict__ out_n,
    int64_t cap) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < nslots; i += stride) {
    uint64_t k = tkeys[i];
    bool take = false;
    int32_t win = 0;
    if (k != EMPTY_SLOT) {
      win = (int32_t)(uint32_t)(k >> 32);
      take = (int64_t)win >= win_lo && (int64_t)win < win_hi;
    }
    unsigned long long ball = __ballot(take);
    int lane = threadIdx.x & (WAVE - 1);
    if (ball != 0) {
      int wave_total = __popcll(ball);
      int rank = __popcll(ball & ((1ULL << lane) - 1ULL));
      int base = 0;
      int lead = __ffsll((unsigned long long)ball) - 1;
      if (lane == lead) base = atomicAdd(out_n, wave_total);
      base = __shfl(base, lead);
      if (take) {
        int64_t idx = base + rank;
        if (idx < cap) {
          out_keys[idx] = (int32_t)(uint32_t)(k & 0xFFFFFFFFULL);
          out_wins[idx] = win;
          out_cnt[idx] = tcnt[i];
          out_sum[idx] = tsum[i];
          out_min[idx] = tmin[i];
          out_max[idx] = tmax[i];
        }
      }
    }
  }
}

// Per-key global join-cell update, "last" insert / "complete" emit
// semantics (reference operators/__init__.py _JoinLogic): store this
// side's value; the update that makes all sides present emits the
// joined row and resets the presence flags, so the next emission
// again requires a fresh value from every side.  Shared by the
// direct insert, the region kernels, and the overflow spill.
__device__ __forceinline__ void join_apply(
    uint64_t packed,
    long long v,
    int side,
    int full,
    uint64_t* __restrict__ tkeys,
    long long* __restrict__ tval0,
    long long* __restrict__ tval1,
    int* __restrict__ tflags,
    uint64_t mask,
    int region_bits,
    int32_t* __restrict__ out_keys,
    int64_t* __restrict__ out_v0,
    int64_t* __restrict__ out_v1,
    int* __restrict__ out_n,
    int64_t out_cap,
    int* __restrict__ error_flag,
    int rearm = 0) {  // the key had >= 2 events this batch: per-item
                      // semantics leave the side flag RE-SET after an
                      // emission (a duplicate lands after the
                      // completing event clears the flags), so a
                      // deduped caller must restore it
  uint64_t slot = find_slot_r(tkeys, mask, region_bits, packed);
  if (slot == ~0ULL) {
    atomicExch(error_flag, 1);
    return;
  }
  if (side == 0) {
    atomicExch((unsigned long long*)&tval0[slot], (unsigned long long)v);
  } else {
    atomicExch((unsigned long long*)&tval1[slot], (unsigned long long)v);
  }
  int old = atomicOr(&tflags[slot], 1 << side);
  if ((old | (1 << side)) == full && old != full) {
    int idx = atomicAdd(out_n, 1);
    if (idx < out_cap) {
      out_keys[idx] = (int32_t)(uint32_t)(packed & 0xFFFFFFFFULL);
      out_v0[idx] = tval0[slot];
      out_v1[idx] = tval1[slot];
    }
    atomicAnd(&tflags[slot], 0);
    if (rearm) atomicOr(&tflags[slot], 1 << side);
  }
}

// Direct (unpartitioned) join insert: one global-table update per
// event.
__global__ void k_join_insert(
    const int32_t* __restrict__ keys,
    const int64_t* __restrict__ vals,
    int64_t n,
    int side,
    int n_sides,
    uint64_t* __restrict__ tkeys,
    long long* __restrict__ tval0,
    long long* __restrict__ tval1,
    int* __restrict__ tflags,
    uint64_t mask,
    int32_t* __restrict__ out_keys,
    int64_t* __restrict__ out_v0,
    int64_t* __restrict__ out_v1,
    int* __restrict__ out_n,
    int64_t cap,
    int* __restrict__ error_flag) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  int full = (1 << n_sides) - 1;
  for (; i < n; i += stride) {
    join_apply((uint64_t)(uint32_t)keys[i], vals[i], side, full, tkeys,
               tval0, tval1, tflags, mask, 0, out_keys, out_v0, out_v1,
               out_n, cap, error_flag);
  }
}

// Region-partitioned join insert: events arrive pre-bucketed into
// per-region segments (k_radix_scatter_fixed with len=2^40 so the
// window id is 0 and `packed` is the bare key); one workgroup per
// region keeps its table slice L2-warm while processing its segment.
__global__ void k_join_region(
    const uint64_t* __restrict__ ev_packed,
    const int64_t* __restrict__ ev_vals,
    const int* __restrict__ counts,
    int64_t cap,
    int side,
    int n_sides,
    uint64_t* __restrict__ tkeys,
    long long* __restrict__ tval0,
    long long* __restrict__ tval1,
    int* __restrict__ tflags,
    uint64_t mask,
    int region_bits,
    int32_t* __restrict__ out_keys,
    int64_t* __restrict__ out_v0,
    int64_t* __restrict__ out_v1,
    int* __restrict__ out_n,
    int64_t out_cap,
    int* __restrict__ error_flag) {
  int b = blockIdx.x;
  int cnt = counts[b];
  if (cnt > (int)cap) cnt = (int)cap;
  int64_t start = (int64_t)b * cap;
  int full = (1 << n_sides) - 1;
  for (int j = threadIdx.x; j < cnt; j += blockDim.x) {
    uint64_t packed = ev_packed[start + j];
    if (packed == EMPTY_SLOT) continue;  // staged-scatter pad
    join_apply(packed, ev_vals[start + j], side, full, tkeys, tval0,
               tval1, tflags, mask, region_bits, out_keys, out_v0,
               out_v1, out_n, out_cap, error_flag);
  }
}

// LDS-deduped region join: one workgroup per region; the region's
// batch segment is first collapsed in LDS to (key -> last value)
// for this side, then merged — ONE global-table update per distinct
// key instead of per event (same staging shape as k_radix_agg).
// Since each launch carries a single side, a per-batch dedup cannot
// change emission counts: at most one not-full -> full transition
// per key per single-side batch either way.
template <int BLK = 256>
__global__ __launch_bounds__(BLK) void k_join_region_lds(
    const uint64_t* __restrict__ ev_packed,
    const int64_t* __restrict__ ev_vals,
    const int* __restrict__ counts,
    int64_t cap,
    int side,
    int n_sides,
    uint64_t* __restrict__ tkeys,
    long long* __restrict__ tval0,
    long long* __restrict__ tval1,
    int* __restrict__ tflags,
    uint64_t mask,
    int region_bits,
    int lds_bits,
    int32_t* __restrict__ out_keys,
    int64_t* __restrict__ out_v0,
    int64_t* __restrict__ out_v1,
    int* __restrict__ out_n,
    int64_t out_cap,
    int* __restrict__ error_flag) {
  extern __shared__ char smem[];
  int R = 1 << lds_bits;
  uint64_t* lkeys = (uint64_t*)smem;
  unsigned long long* lvals =
      (unsigned long long*)(smem + (size_t)R * sizeof(uint64_t));
  int* lcnt = (int*)(smem + (size_t)R * 2 * sizeof(uint64_t));
  for (int s = threadIdx.x; s < R; s += blockDim.x) {
    lkeys[s] = EMPTY_SLOT;
    lcnt[s] = 0;
  }
  __syncthreads();
  int b = blockIdx.x;
  int cnt = counts[b];
  if (cnt > (int)cap) cnt = (int)cap;
  int64_t start = (int64_t)b * cap;
  int full = (1 << n_sides) - 1;
  for (int j = threadIdx.x; j < cnt; j += blockDim.x) {
    uint64_t packed = ev_packed[start + j];
    if (packed == EMPTY_SLOT) continue;  // staged-scatter pad
    unsigned long long v = (unsigned long long)ev_vals[start + j];
    uint64_t h64 = mix64(packed);
    int lh = (int)((h64 >> 32) & (R - 1));
    bool done = false;
    for (int p = 0; p < R; ++p) {
      uint64_t cur = lkeys[lh];
      if (cur == packed) {
        atomicExch(&lvals[lh], v);
        atomicAdd(&lcnt[lh], 1);
        done = true;
        break;
      }
      if (cur == EMPTY_SLOT) {
        uint64_t prev = atomicCAS(
            (unsigned long long*)&lkeys[lh], EMPTY_SLOT, packed);
        if (prev == EMPTY_SLOT || prev == packed) {
          atomicExch(&lvals[lh], v);
          atomicAdd(&lcnt[lh], 1);
          done = true;
          break;
        }
      }
      lh = (lh + 1) & (R - 1);
    }
    if (!done) {
      // LDS staging full: apply straight to the global region.
      join_apply(packed, (long long)v, side, full, tkeys, tval0, tval1,
                 tflags, mask, region_bits, out_keys, out_v0, out_v1,
                 out_n, out_cap, error_flag);
    }
  }
  __syncthreads();
  // Merge distinct keys once into the global table (the segment's
  // region is contiguous in HBM, so the probes stay local); keys
  // with >= 2 events keep per-item re-arm semantics via join_apply.
  for (int s = threadIdx.x; s < R; s += blockDim.x) {
    if (lkeys[s] != EMPTY_SLOT) {
      join_apply(lkeys[s], (long long)lvals[s], side, full, tkeys,
                 tval0, tval1, tflags, mask, region_bits, out_keys,
                 out_v0, out_v1, out_n, out_cap, error_flag,
                 lcnt[s] >= 2);
    }
  }
}

// Overflow spill for the region join (direct path).
__global__ void k_join_overflow(
    const uint64_t* __restrict__ ov_packed,
    const int64_t* __restrict__ ov_vals,
    const int* __restrict__ ov_cursor,
    int64_t ov_cap,
    int side,
    int n_sides,
    uint64_t* __restrict__ tkeys,
    long long* __restrict__ tval0,
    long long* __restrict__ tval1,
    int* __restrict__ tflags,
    uint64_t mask,
    int region_bits,
    int32_t* __restrict__ out_keys,
    int64_t* __restrict__ out_v0,
    int64_t* __restrict__ out_v1,
    int* __restrict__ out_n,
    int64_t out_cap,
    int* __restrict__ error_flag) {
  int64_t n = *ov_cursor;
  if (n > ov_cap) n = ov_cap;
  int full = (1 << n_sides) - 1;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    join_apply(ov_packed[i], ov_vals[i], side, full, tkeys, tval0,
               tval1, tflags, mask, region_bits, out_keys, out_v0,
               out_v1, out_n, out_cap, error_flag);
  }
}

// Extract live join-side state (for recovery snapshots).
__global__ void k_join_extract(
    uint64_t* __restrict__ tkeys,
    long long* __restrict__ tval0,
    long long* __restrict__ tval1,
    int* __restrict__ tflags,
    int64_t nslots,
    int32_t* __restrict__ out_keys,
    int64_t* __restrict__ out_v0,
    int64_t* __restrict__ out_v1,
    int32_t* __restrict__ out_flags,
    int* __restrict__ out_n,
    int64_t cap) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < nslots; i += stride) {
    uint64_t k = tkeys[i];
    bool take = (k != EMPTY_SLOT) && (tflags[i] != 0);
    unsigned long long ball = __ballot(take);
    int lane = threadIdx.x & (WAVE - 1);
    if (ball != 0) {
      int wave_total = __popcll(ball);
      int rank = __popcll(ball & ((1ULL << lane) - 1ULL));
      int base = 0;
      int lead = __ffsll((unsigned long long)ball) - 1;
      if (lane == lead) base = atomicAdd(out_n, wave_total);
      base = __shfl(base, lead);
      if (take) {
        int64_t idx = base + rank;
        if (idx < cap) {
          out_keys[idx] = (int32_t)(uint32_t)(k & 0xFFFFFFFFULL);
          out_v0[idx] = tval0[i];
          out_v1[idx] = tval1[i];
          out_flags[idx] = tflags[i];
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Session windows (gap-based).  Per-key state is one table cell:
// (session_start, last_ts, accumulator).  Batches arrive SORTED by
// (key, ts) — the host wrapper sorts — and one thread walks each
// key's contiguous segment in order: a gap > gap_ms closes the
// running session (emit) and starts a new one.  Parallelism is the
// distinct-key count of the batch, which is the right shape for the
// high-cardinality workloads the device path targets (low-cardinality
// session streams belong on the host path).
//
// Reference semantics: bytewax windowing.py _SessionWindowerLogic
// (gap merge); here in-order ingestion makes merges degenerate to
// extension, matching the reference under watermark-ordered input.

// Table layout: skeys int64 (EMPTY_SLOT = empty), svals 3x int64 per
// slot (start, last, acc).
__device__ inline uint64_t session_find_or_claim(
    uint64_t* skeys, uint64_t mask, uint64_t key) {
  uint64_t h = mix64(key);
  for (uint64_t probe = 0; probe <= mask; ++probe) {
    uint64_t slot = (h + probe) & mask;
    uint64_t cur = skeys[slot];
    if (cur == key) return slot;
    if (cur == EMPTY_SLOT) {
      uint64_t prev = atomicCAS(
          (unsigned long long*)&skeys[slot], EMPTY_SLOT, key);
      if (prev == EMPTY_SLOT || prev == key) return slot;
    }
  }
  return ~0ULL;
}

template <int MODE>
__global__ void k_session_insert(
    const int32_t* __restrict__ keys,   // sorted by (key, ts)
    const int64_t* __restrict__ ts,
    const int64_t* __restrict__ vals,
    const int64_t* __restrict__ seg_start,  // [n_segs] segment offsets
    const int64_t* __restrict__ seg_end,
    int64_t n_segs,
    int64_t gap_ms,
    uint64_t* __restrict__ skeys,
    long long* __restrict__ sstart,
    long long* __restrict__ slast,
    long long* __restrict__ sacc,
    uint64_t mask,
    int32_t* __restrict__ out_keys,  // closed sessions
    int64_t* __restrict__ out_start,
    int64_t* __restrict__ out_end,
    int64_t* __restrict__ out_vals,
    int* __restrict__ out_n,
    int64_t out_cap,
    unsigned long long* __restrict__ max_ts,
    int* __restrict__ error_flag) {
  int64_t seg = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; seg < n_segs; seg += stride) {
    int64_t i = seg_start[seg];
    int64_t end = seg_end[seg];
    if (i >= end) continue;
    uint64_t key = (uint64_t)(uint32_t)keys[i];
    uint64_t slot = session_find_or_claim(skeys, mask, key);
    if (slot == ~0ULL) {
      atomicExch(error_flag, 1);
      continue;
    }
    // This thread owns the key's whole segment; no other thread
    // touches this slot within the launch.
    long long start = sstart[slot];
    long long last = slast[slot];
    long long acc = sacc[slot];
    bool open = last >= 0;
    for (; i < end; ++i) {
      int64_t t = ts[i];
      int64_t v = (MODE == AGG_SUM) ? vals[i] : 1;
      if (open && t - last > gap_ms) {
        int idx = atomicAdd(out_n, 1);
        if (idx < out_cap) {
          out_keys[idx] = (int32_t)(uint32_t)key;
          out_start[idx] = start;
          out_end[idx] = last;
          out_vals[idx] = acc;
        } else {
          atomicExch(error_flag, 1);
        }
        open = false;
      }
      if (!open) {
        start = t;
        acc = 0;
        open = true;
      }
      last = t;
      acc += v;
    }
    sstart[slot] = start;
    slast[slot] = last;
    sacc[slot] = acc;
    for (int off = WAVE / 2; off > 0; off >>= 1) {
      long long other = __shfl_down(last, off);
      if (other > last) last = other;
    }
    if ((threadIdx.x & (WAVE - 1)) == 0) {
      atomicMax(max_ts, (unsigned long long)last);
    }
  }
}

// Close sessions whose last event is older than `horizon_ms` (the
// watermark minus the gap): emit and clear the cell.  Clearing is
// safe here, unlike the windowed table, because the NEXT insert
// re-claims slots via CAS (session_find_or_claim tolerates probe
// holes by claiming the first empty slot; a key displaced past a
// cleared hole simply occupies two probes' worth of distance — the
// chain is never read without the CAS claim).  To keep lookups exact
// we still migrate live cells to the alternate table, mirroring the
// windowed close.
__global__ void k_session_close_migrate(
    const uint64_t* __restrict__ skeys,
    const long long* __restrict__ sstart,
    const long long* __restrict__ slast,
    const long long* __restrict__ sacc,
    int64_t nslots,
    int64_t horizon_ms,
    uint64_t* __restrict__ dst_keys,
    long long* __restrict__ dst_start,
    long long* __restrict__ dst_last,
    long long* __restrict__ dst_acc,
    uint64_t mask,
    int32_t* __restrict__ out_keys,
    int64_t* __restrict__ out_start,
    int64_t* __restrict__ out_end,
    int64_t* __restrict__ out_vals,
    int* __restrict__ out_n,
    int64_t out_cap,
    int* __restrict__ error_flag) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < nslots; i += stride) {
    uint64_t key = skeys[i];
    if (key == EMPTY_SLOT) continue;
    long long last = slast[i];
    if (last < 0) continue;
    if (last < horizon_ms) {
      int idx = atomicAdd(out_n, 1);
      if (idx < out_cap) {
        out_keys[idx] = (int32_t)(uint32_t)key;
        out_start[idx] = sstart[i];
        out_end[idx] = last;
        out_vals[idx] = sacc[i];
      } else {
        atomicExch(error_flag, 1);
      }
    } else {
      uint64_t slot = session_find_or_claim(dst_keys, mask, key);
      if (slot == ~0ULL) {
        atomicExch(error_flag, 1);
        continue;
      }
      dst_start[slot] = sstart[i];
      dst_last[slot] = last;
      dst_acc[slot] = sacc[i];
    }
  }
}

// Stream compaction: keep events where mask != 0, preserving relative
// order per wave (wave-ballot ranks + one atomic per wave).
__global__ void k_filter_compact(
    const int32_t* __restrict__ keys,
    const int64_t* __restrict__ ts,
    const int64_t* __restrict__ vals,  // may be nullptr
    const uint8_t* __restrict__ mask,
    int64_t n,
    int32_t* __restrict__ out_keys,
    int64_t* __restrict__ out_ts,
    int64_t* __restrict__ out_vals,
    int* __restrict__ out_n) {
  int lane = threadIdx.x & (WAVE - 1);
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  int64_t first = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  for (int64_t i = first; i - lane < n; i += stride) {
    bool keep = (i < n) && mask[i] != 0;
    unsigned long long ball = __ballot(keep);
    if (ball == 0) continue;
    int wave_total = __popcll(ball);
    int rank = __popcll(ball & ((1ULL << lane) - 1ULL));
    int base = 0;
    int lead = __ffsll(ball) - 1;
    if (lane == lead) base = atomicAdd(out_n, wave_total);
    base = __shfl(base, lead);
    if (keep) {
      int64_t o = base + rank;
      out_keys[o] = keys[i];
      out_ts[o] = ts[i];
      if (vals != nullptr) out_vals[o] = vals[i];
    }
  }
}

// Histogram of destination workers for the keyed exchange.
__global__ void k_bucket_hist(
    const int32_t* __restrict__ keys,
    int64_t n,
    int world,
    int* __restrict__ counts) {
  extern __shared__ int lcounts[];
  for (int w = threadIdx.x; w < world; w += blockDim.x) lcounts[w] = 0;
  __syncthreads();
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    int dst = (int)(mix64((uint32_t)keys[i]) % (uint64_t)world);
    atomicAdd(&lcounts[dst], 1);
  }
  __syncthreads();
  for (int w = threadIdx.x; w < world; w += blockDim.x) {
    if (lcounts[w] > 0) atomicAdd(&counts[w], lcounts[w]);
  }
}

// Scatter events into per-destination contiguous segments.
// `cursors` must be pre-loaded with the exclusive prefix sums of the
// destination counts.  Order within a destination is not preserved
// (items within an epoch are unordered).
template <typename TS = int64_t>
__global__ void k_bucket_scatter(
    const int32_t* __restrict__ keys,
    const TS* __restrict__ ts,
    const int64_t* __restrict__ vals,  // may be nullptr
    int64_t n,
    int world,
    int* __restrict__ cursors,
    int32_t* __restrict__ out_keys,
    TS* __restrict__ out_ts,
    int64_t* __restrict__ out_vals) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    int32_t k = keys[i];
    int dst = (int)(mix64((uint32_t)k) % (uint64_t)world);
    int idx = atomicAdd(&cursors[dst], 1);
    out_keys[idx] = k;
    out_ts[idx] = ts[i];
    if (vals != nullptr) out_vals[idx] = vals[i];
  }
}

// ---------------------------------------------------------------------------
// String dictionary encode (str keys -> dense int32 ids, on device).
//
// The reference's key contract is `str` (reference src/operators.rs:
// 363-439 extract_key); the columnar fast path keys RecordBatches by
// int32 id.  These kernels bridge the two: string bytes (+offsets)
// land on device once, each string is 128-bit hashed and mapped to a
// dense id through an open-address device dictionary.
//
// Exactness: the device table stores the full 128-bit hash (lo is
// the CAS claim word, hi the published check word).  Newly created
// ids are read back as (id, batch_index) pairs so the host keeps the
// authoritative id->string list; the Python layer verifies each new
// id against its own exact dict and fails loudly on a 128-bit
// collision (probability ~1e-26 at 1e6 keys) instead of aggregating
// wrong.
//
// Race-free without spins: insert and lookup are SEPARATE kernel
// launches.  k_dict_insert only guarantees every distinct string has
// a claimed slot with published (hi, id) by kernel end (threads that
// lose the CAS to the same hash exit; the winner's hi/id stores are
// globally visible at the kernel boundary).  k_dict_lookup then
// resolves every event's id with plain loads.
// ---------------------------------------------------------------------------

__device__ __forceinline__ void str_hash128(
    const uint8_t* __restrict__ s, int len, uint64_t* h_lo,
    uint64_t* h_hi) {
  // FNV-1a over the bytes, finalized twice with independent seeds.
  uint64_t h = 0xcbf29ce484222325ULL;
  for (int i = 0; i < len; ++i) {
    h ^= (uint64_t)s[i];
    h *= 0x100000001b3ULL;
  }
  uint64_t lo = mix64(h ^ 0x9e3779b97f4a7c15ULL);
  if (lo == EMPTY_SLOT) lo = 0;  // reserve the empty sentinel
  *h_lo = lo;
  *h_hi = mix64(h + 0x2545f4914f6cdd1dULL);
}

__global__ void k_dict_insert(
    const uint8_t* __restrict__ bytes,
    const int64_t* __restrict__ offs,  // [n+1]
    int64_t n,
    uint64_t* __restrict__ dlo,   // [nslots] CAS claim words
    uint64_t* __restrict__ dhi,   // [nslots] published check words
    int32_t* __restrict__ dids,   // [nslots] published ids
    uint64_t mask,
    int* __restrict__ counter,        // next id
    int32_t* __restrict__ new_ids,    // [new_cap] readback: id
    int32_t* __restrict__ new_idx,    // [new_cap] readback: batch index
    int* __restrict__ new_n,
    int64_t new_cap,
    int* __restrict__ error_flag) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    int len = (int)(offs[i + 1] - offs[i]);
    uint64_t lo, hi;
    str_hash128(bytes + offs[i], len, &lo, &hi);
    uint64_t h = lo & mask;
    for (uint64_t probes = 0; probes <= mask; ++probes) {
      uint64_t cur = dlo[h];
      if (cur == lo) break;  // claimed (here or elsewhere): winner publishes
      if (cur == EMPTY_SLOT) {
        uint64_t prev = atomicCAS((unsigned long long*)&dlo[h],
                                  EMPTY_SLOT, lo);
        if (prev == EMPTY_SLOT) {
          int id = atomicAdd(counter, 1);
          dhi[h] = hi;
          dids[h] = id;
          int r = atomicAdd(new_n, 1);
          if (r < new_cap) {
            new_ids[r] = id;
            new_idx[r] = (int32_t)i;
          } else {
            atomicExch(error_flag, 3);  // readback buffer overflow
          }
          break;
        }
        if (prev == lo) break;
      }
      h = (h + 1) & mask;
      if (probes == mask) atomicExch(error_flag, 1);  // table full
    }
  }
}

__global__ void k_dict_lookup(
    const uint8_t* __restrict__ bytes,
    const int64_t* __restrict__ offs,
    int64_t n,
    const uint64_t* __restrict__ dlo,
    const uint64_t* __restrict__ dhi,
    const int32_t* __restrict__ dids,
    uint64_t mask,
    int32_t* __restrict__ out_ids,
    int* __restrict__ error_flag) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    int len = (int)(offs[i + 1] - offs[i]);
    uint64_t lo, hi;
    str_hash128(bytes + offs[i], len, &lo, &hi);
    uint64_t h = lo & mask;
    int32_t id = -1;
    for (uint64_t probes = 0; probes <= mask; ++probes) {
      uint64_t cur = dlo[h];
      if (cur == lo && dhi[h] == hi) {
        id = dids[h];
        break;
      }
      if (cur == EMPTY_SLOT) break;
      h = (h + 1) & mask;
    }
    out_ids[i] = id;
    if (id < 0) atomicExch(error_flag, 2);  // unpublished / collision
  }
}

// Restore path: re-claim slots with PINNED ids (snapshot order must
// be reproduced exactly across restarts).
__global__ void k_dict_insert_pinned(
    const uint8_t* __restrict__ bytes,
    const int64_t* __restrict__ offs,
    const int32_t* __restrict__ ids,
    int64_t n,
    uint64_t* __restrict__ dlo,
    uint64_t* __restrict__ dhi,
    int32_t* __restrict__ dids,
    uint64_t mask,
    int* __restrict__ error_flag) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    int len = (int)(offs[i + 1] - offs[i]);
    uint64_t lo, hi;
    str_hash128(bytes + offs[i], len, &lo, &hi);
    uint64_t h = lo & mask;
    for (uint64_t probes = 0; probes <= mask; ++probes) {
      uint64_t prev = atomicCAS((unsigned long long*)&dlo[h],
                                EMPTY_SLOT, lo);
      if (prev == EMPTY_SLOT) {
        dhi[h] = hi;
        dids[h] = ids[i];
        break;
      }
      h = (h + 1) & mask;
      if (probes == mask) atomicExch(error_flag, 1);
    }
  }
}

// ---------------------------------------------------------------------------
// String exchange: route raw string bytes to their owning rank by
// CONTENT hash, so multi-GPU str-keyed streams encode at the owner
// and dictionary ids never cross ranks (bytewax_amd/gpu/strings.py).
// Wire order is bucket-major: lengths/ts/vals in per-destination
// contiguous segments, bytes packed in the same relative order
// (receivers rebuild offsets with a cumsum over lengths).
// ---------------------------------------------------------------------------

__device__ __forceinline__ int str_owner(
    const uint8_t* __restrict__ bytes, const int64_t* __restrict__ offs,
    int64_t i, int world) {
  uint64_t lo, hi;
  str_hash128(bytes + offs[i], (int)(offs[i + 1] - offs[i]), &lo, &hi);
  return (int)(hi % (uint64_t)world);
}

__global__ void k_str_bucket_hist(
    const uint8_t* __restrict__ bytes,
    const int64_t* __restrict__ offs,
    int64_t n,
    int world,
    int* __restrict__ counts,            // [world] strings
    long long* __restrict__ byte_counts  // [world] bytes
) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    int dst = str_owner(bytes, offs, i, world);
    atomicAdd(&counts[dst], 1);
    atomicAdd((unsigned long long*)&byte_counts[dst],
              (unsigned long long)(offs[i + 1] - offs[i]));
  }
}

__global__ void k_str_meta_scatter(
    const uint8_t* __restrict__ bytes,
    const int64_t* __restrict__ offs,
    const int64_t* __restrict__ ts,
    const int64_t* __restrict__ vals,  // may be nullptr
    int64_t n,
    int world,
    int* __restrict__ cursors,  // [world] exclusive prefix of counts
    int32_t* __restrict__ send_lens,
    int64_t* __restrict__ send_ts,
    int64_t* __restrict__ send_vals,
    int32_t* __restrict__ src_idx) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    int dst = str_owner(bytes, offs, i, world);
    int pos = atomicAdd(&cursors[dst], 1);
    send_lens[pos] = (int32_t)(offs[i + 1] - offs[i]);
    send_ts[pos] = ts[i];
    if (vals != nullptr) send_vals[pos] = vals[i];
    src_idx[pos] = (int32_t)i;
  }
}

__global__ void k_str_byte_gather(
    const uint8_t* __restrict__ bytes,
    const int64_t* __restrict__ offs,
    const int32_t* __restrict__ src_idx,
    const int64_t* __restrict__ send_offs,  // [n] exclusive cumsum of lens
    int64_t n,
    uint8_t* __restrict__ send_bytes) {
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; j < n; j += stride) {
    int64_t i = src_idx[j];
    int64_t src = offs[i];
    int len = (int)(offs[i + 1] - src);
    int64_t dst = send_offs[j];
    for (int b = 0; b < len; ++b) send_bytes[dst + b] = bytes[src + b];
  }
}

// ---------------------------------------------------------------------------
// Host-side wrappers (torch extension API)
// ---------------------------------------------------------------------------

static void check_dev(const torch::Tensor& t, torch::ScalarType st,
                      const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a device tensor");
  TORCH_CHECK(t.scalar_type() == st, name, " has wrong dtype");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

void window_agg_insert(
    torch::Tensor keys,
    torch::Tensor ts,
    c10::optional<torch::Tensor> vals,
    torch::Tensor tkeys,
    torch::Tensor tvals,
    torch::Tensor max_ts,
    torch::Tensor error_flag,
    int64_t align_ms,
    int64_t len_ms,
    int64_t mode,
    bool dedup,
    int64_t ts_base,
    int64_t region_bits,
    int64_t off_ms) {
  if (off_ms <= 0) off_ms = len_ms;
  TORCH_CHECK(off_ms <= len_ms, "window offset must be <= length");
  TORCH_CHECK(off_ms == len_ms || !dedup,
              "sliding windows are incompatible with the dedup path");
  check_dev(keys, torch::kInt32, "keys");
  bool ts32 = ts.scalar_type() == torch::kInt32;
  if (!ts32) check_dev(ts, torch::kInt64, "ts");
  check_dev(tkeys, torch::kInt64, "tkeys");
  check_dev(tvals, torch::kInt64, "tvals");
  check_dev(max_ts, torch::kInt64, "max_ts");
  check_dev(error_flag, torch::kInt32, "error_flag");
  int64_t n = keys.numel();
  TORCH_CHECK(ts.numel() == n, "ts/keys length mismatch");
  int64_t nslots = tkeys.numel();
  TORCH_CHECK((nslots & (nslots - 1)) == 0, "table size must be 2^k");
  TORCH_CHECK(len_ms > 0, "window length must be positive");
  const int64_t* vptr = nullptr;
  if (mode == AGG_SUM) {
    TORCH_CHECK(vals.has_value(), "sum mode requires vals");
    check_dev(*vals, torch::kInt64, "vals");
    TORCH_CHECK(vals->numel() == n, "vals/keys length mismatch");
    vptr = vals->data_ptr<int64_t>();
  }
  if (n == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(n, 256));
  auto launch = [&](auto kern, auto tsptr) {
    hipLaunchKernelGGL(
        kern, grid, block, 0, stream,
        keys.data_ptr<int32_t>(), tsptr, vptr, n,
        (uint64_t*)tkeys.data_ptr<int64_t>(),
        (unsigned long long*)tvals.data_ptr<int64_t>(),
        (uint64_t)(nslots - 1), align_ms, len_ms, off_ms, ts_base,
        (int)region_bits,
        (unsigned long long*)max_ts.data_ptr<int64_t>(),
        error_flag.data_ptr<int32_t>(), (const int64_t*)nullptr);
  };
  auto dispatch = [&](auto tsptr) {
    using TSV = std::remove_const_t<std::remove_pointer_t<decltype(tsptr)>>;
    if (mode == AGG_COUNT && !dedup)
      launch(k_window_agg_insert<AGG_COUNT, false, TSV>, tsptr);
    else if (mode == AGG_COUNT && dedup)
      launch(k_window_agg_insert<AGG_COUNT, true, TSV>, tsptr);
    else if (mode == AGG_SUM && !dedup)
      launch(k_window_agg_insert<AGG_SUM, false, TSV>, tsptr);
    else
      launch(k_window_agg_insert<AGG_SUM, true, TSV>, tsptr);
  };
  if (ts32) dispatch(ts.data_ptr<int32_t>());
  else dispatch(ts.data_ptr<int64_t>());
}

void radix_window_insert(
    torch::Tensor keys,
    torch::Tensor ts,
    c10::optional<torch::Tensor> vals,
    torch::Tensor tkeys,
    torch::Tensor tvals,
    torch::Tensor max_ts,
    torch::Tensor error_flag,
    torch::Tensor gcursors,   // int32 [n_regions]
    torch::Tensor ev_packed,  // int64 [n_regions * cap]
    torch::Tensor ev_vals,    // int64 [n_regions * cap] (sum mode)
    torch::Tensor ov_cursor,  // int32 [1]
    torch::Tensor ov_packed,  // int64 overflow spill
    torch::Tensor ov_vals,
    int64_t align_ms,
    int64_t len_ms,
    int64_t mode,
    int64_t ts_base,
    int64_t region_bits,
    int64_t off_ms = 0,  // sliding stride; 0 or == len_ms for tumbling
    std::vector<int64_t> seg_counts = {},
    std::vector<int64_t> seg_bases = {}) {
  if (off_ms <= 0) off_ms = len_ms;
  TORCH_CHECK(off_ms <= len_ms, "window offset must be <= length");
  check_dev(keys, torch::kInt32, "keys");
  // Segmented form: `ts` holds int32 deltas laid out as contiguous
  // per-source-rank segments (the RCCL all-to-allv wire format);
  // seg_counts[i] events use seg_bases[i] as their absolute base.
  // This inserts exchange output without ever materializing int64
  // timestamps (saves ~3 HBM passes over the biggest column).
  bool seg32 = !seg_counts.empty();
  bool ts32 = ts.scalar_type() == torch::kInt32;
  if (seg32) {
    TORCH_CHECK(seg_counts.size() == seg_bases.size(),
                "seg_counts/seg_bases length mismatch");
    check_dev(ts, torch::kInt32, "ts32");
  } else if (!ts32) {
    check_dev(ts, torch::kInt64, "ts");
  }
  int64_t n = keys.numel();
  int64_t nslots = tkeys.numel();
  TORCH_CHECK((nslots & (nslots - 1)) == 0, "table size must be 2^k");
  TORCH_CHECK(region_bits > 0, "radix path requires region_bits > 0");
  TORCH_CHECK(region_bits <= 12, "region_bits > 12 exceeds the LDS "
              "budget of the aggregation kernel (16 B/slot)");
  int64_t nb = nslots >> region_bits;
  TORCH_CHECK(nb >= 1 && nb <= 8192, "region count out of range");
  TORCH_CHECK(gcursors.numel() >= nb, "gcursors too small");
  const int64_t* vptr = nullptr;
  if (mode == AGG_SUM) {
    TORCH_CHECK(vals.has_value(), "sum mode requires vals");
    vptr = vals->data_ptr<int64_t>();
  }
  if (n == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  uint64_t mask = (uint64_t)(nslots - 1);
  dim3 block(256);
  dim3 grid(n_blocks(n, 256));

  ScatterKind kind = scatter_kind_env();
  // An event expands into ceil(len/off) windows when sliding; the
  // direct variant has no expansion loop.
  int64_t xf = off_ms < len_ms ? (len_ms + off_ms - 1) / off_ms : 1;
  TORCH_CHECK(xf <= 64, "sliding expansion > 64x; use a larger offset");
  if (kind == SCAT_DIRECT && xf > 1) kind = SCAT_FIXED;
  int coarse = scatter_coarse_bits(kind);
  int seg_bits = (int)region_bits + coarse;
  int64_t nseg = nslots >> seg_bits;
  // LDS budgets: staged scatter stages SC_GRAN packed (+vals) per
  // segment; the agg kernel stages 16 B per LDS slot.
  size_t staged_lds = (size_t)nseg * SC_GRAN * 8 *
                          (mode == AGG_SUM ? 2 : 1) +
                      4 * (size_t)nseg * sizeof(int);
  while (kind == SCAT_STAGED &&
         (nseg < 1 || staged_lds > 160 * 1024 || seg_bits > 13)) {
    if (coarse > 0 && seg_bits > 13) {
      coarse -= 1;
    } else {
      kind = SCAT_FIXED;
      coarse = 0;
    }
    seg_bits = (int)region_bits + coarse;
    nseg = nslots >> seg_bits;
    staged_lds = (size_t)nseg * SC_GRAN * 8 * (mode == AGG_SUM ? 2 : 1) +
                 4 * (size_t)nseg * sizeof(int);
  }
  if (seg_bits > 13) {
    coarse = 0;
    seg_bits = (int)region_bits;
    nseg = nb;
  }
  int64_t cap = ev_packed.numel() / nseg;
  if (kind == SCAT_STAGED) {
    cap &= ~(int64_t)(SC_GRAN - 1);
    if (cap < SC_GRAN) {
      // Tiny buffers (small batches): granule rounding would zero the
      // per-segment capacity; the fixed variant has no granularity.
      kind = SCAT_FIXED;
      coarse = 0;
      seg_bits = (int)region_bits;
      nseg = nb;
      cap = ev_packed.numel() / nseg;
    }
  }
  TORCH_CHECK(cap * nseg >= 2 * n * xf || cap >= n * xf,
              "scatter buffers too small (need ~2x batch x expansion)");
  if (mode == AGG_SUM) {
    TORCH_CHECK(ev_vals.numel() >= nseg * cap, "ev_vals too small");
  }
  gcursors.narrow(0, 0, nseg).zero_();
  ov_cursor.zero_();
  size_t hist_lds = (size_t)nseg * sizeof(int);

  // Scatter grid: fewer blocks keep fewer segment cursors (and thus
  // partially-written cache lines) open at once — per-XCD L2 can then
  // accumulate full lines before eviction.  Tunable while we converge
  // on the best default (BYTEWAX_SCATTER_BLOCKS).
  int env_blocks = 0;
  if (const char* sb = std::getenv("BYTEWAX_SCATTER_BLOCKS")) {
    int v = atoi(sb);
    if (v > 0) env_blocks = v;
  }

  struct Seg {
    int64_t off, n, base;
  };
  std::vector<Seg> segs;
  if (seg32) {
    int64_t off = 0;
    for (size_t i = 0; i < seg_counts.size(); ++i) {
      if (seg_counts[i] > 0) segs.push_back({off, seg_counts[i], seg_bases[i]});
      off += seg_counts[i];
    }
    TORCH_CHECK(off == n, "segment counts must sum to batch length");
  } else {
    segs.push_back({0, n, ts_base});
  }

  uint64_t win_m2, win_maxfast;
  magic_div_u64(off_ms, &win_m2, &win_maxfast);
  unsigned scat_threads = 256;
  auto scat = [&](auto kern, auto tsptr, const Seg& sg, unsigned gx,
                  size_t lds) {
    hipLaunchKernelGGL(
        kern, dim3(gx), dim3(scat_threads), lds, stream,
        keys.data_ptr<int32_t>() + sg.off, tsptr + sg.off,
        vptr != nullptr ? vptr + sg.off : nullptr, sg.n, align_ms, len_ms,
        off_ms, sg.base, mask, seg_bits, cap, gcursors.data_ptr<int32_t>(),
        (uint64_t*)ev_packed.data_ptr<int64_t>(),
        mode == AGG_SUM ? ev_vals.data_ptr<int64_t>() : nullptr,
        ov_cursor.data_ptr<int32_t>(),
        (uint64_t*)ov_packed.data_ptr<int64_t>(),
        mode == AGG_SUM ? ov_vals.data_ptr<int64_t>() : nullptr,
        ov_packed.numel(),
        (unsigned long long*)max_ts.data_ptr<int64_t>(),
        error_flag.data_ptr<int32_t>(), win_m2, win_maxfast);
  };
  auto scat_any = [&](auto tsptr, const Seg& sg) {
    using TSV = std::remove_const_t<std::remove_pointer_t<decltype(tsptr)>>;
    unsigned gx = (unsigned)n_blocks(sg.n, 256);
    if (env_blocks > 0 && (unsigned)env_blocks < gx) gx = (unsigned)env_blocks;
    if (kind == SCAT_STAGED) {
      // One tile = blockDim * U events; enough blocks to fill the
      // chip at the LDS-bounded occupancy, few enough to keep the
      // end-of-kernel residual padding small.  U (events per thread
      // per tile) tunable via BYTEWAX_SCATTER_U (8/16/32).
      int su = 16;
      if (const char* e = std::getenv("BYTEWAX_SCATTER_U")) {
        int v = atoi(e);
        if (v == 8 || v == 16 || v == 32) su = v;
      }
      // 512-thread workgroups measured fastest for the COUNT staged
      // scatter (60.1 vs 56.6e9 at 256, 57.2 at 1024 — see profiles/).
      int sthreads = mode == AGG_COUNT ? 512 : 256;
      if (const char* e = std::getenv("BYTEWAX_SCATTER_THREADS")) {
        int v = atoi(e);
        if (v == 256 || v == 512 || v == 1024) sthreads = v;
      }
      scat_threads = (unsigned)sthreads;
      unsigned cap_gs = env_blocks > 0 ? (unsigned)env_blocks : 1024u;
      unsigned gs =
          (unsigned)((sg.n * xf + sthreads * su - 1) / (sthreads * su));
      if (gs > cap_gs) gs = cap_gs;
      if (gs < 1) gs = 1;
      size_t s2_lds = (size_t)(256 * 16 + 7 * nseg) * 12 +
                      (size_t)nseg * SC_GRAN * 8 +
                      7 * (size_t)nseg * sizeof(int) + 16;
      if (mode == AGG_COUNT && xf == 1 && scatter_staged2_env() &&
          s2_lds <= 160 * 1024) {
        scat_threads = 256;
        scat(k_radix_scatter_staged2<TSV>, tsptr, sg, gs, s2_lds);
      } else if (mode == AGG_COUNT) {
        if (sthreads == 512)
          scat((k_radix_scatter_staged<AGG_COUNT, TSV, 16, 512>), tsptr,
               sg, gs, staged_lds);
        else if (sthreads == 1024)
          scat((k_radix_scatter_staged<AGG_COUNT, TSV, 16, 1024>), tsptr,
               sg, gs, staged_lds);
        else if (su == 8)
          scat(k_radix_scatter_staged<AGG_COUNT, TSV, 8>, tsptr, sg, gs,
               staged_lds);
        else if (su == 32)
          scat(k_radix_scatter_staged<AGG_COUNT, TSV, 32>, tsptr, sg, gs,
               staged_lds);
        else
          scat(k_radix_scatter_staged<AGG_COUNT, TSV, 16>, tsptr, sg, gs,
               staged_lds);
      } else {
        scat_threads = 256;
        scat(k_radix_scatter_staged<AGG_SUM, TSV>, tsptr, sg, gs,
             staged_lds);
      }
    } else if (kind == SCAT_DIRECT) {
      scat_threads = 256;
      if (mode == AGG_COUNT)
        scat(k_radix_scatter_direct<AGG_COUNT, TSV>, tsptr, sg, gx, 0);
      else
        scat(k_radix_scatter_direct<AGG_SUM, TSV>, tsptr, sg, gx, 0);
    } else {
      scat_threads = 256;
      if (mode == AGG_COUNT)
        scat(k_radix_scatter_fixed<AGG_COUNT, TSV>, tsptr, sg, gx,
             2 * hist_lds);
      else
        scat(k_radix_scatter_fixed<AGG_SUM, TSV>, tsptr, sg, gx,
             2 * hist_lds);
    }
  };
  for (const Seg& sg : segs) {
    if (seg32 || ts32) scat_any(ts.data_ptr<int32_t>(), sg);
    else scat_any(ts.data_ptr<int64_t>(), sg);
  }

  // Fixed layout: segment b's events live at [b*cap, b*cap + count).
  auto offsets = at::arange(
      nseg, at::TensorOptions().dtype(at::kInt).device(keys.device()));
  offsets = offsets * (int)cap;
  size_t agg_lds = (size_t)16 << seg_bits;
  dim3 agg_block(seg_bits >= 12 ? 1024 : 256);
  auto agg = [&](auto kern) {
    hipLaunchKernelGGL(
        kern, dim3((unsigned)nseg), agg_block, agg_lds, stream,
        (const uint64_t*)ev_packed.data_ptr<int64_t>(),
        mode == AGG_SUM ? ev_vals.data_ptr<int64_t>() : nullptr,
        offsets.data_ptr<int32_t>(), gcursors.data_ptr<int32_t>(), cap,
        (uint64_t*)tkeys.data_ptr<int64_t>(),
        (unsigned long long*)tvals.data_ptr<int64_t>(), mask,
        (int)region_bits, seg_bits, error_flag.data_ptr<int32_t>());
  };
  if (mode == AGG_COUNT) agg(k_radix_agg<AGG_COUNT>);
  else agg(k_radix_agg<AGG_SUM>);

  auto ov = [&](auto kern) {
    hipLaunchKernelGGL(
        kern, dim3(64), dim3(256), 0, stream,
        (const uint64_t*)ov_packed.data_ptr<int64_t>(),
        mode == AGG_SUM ? ov_vals.data_ptr<int64_t>() : nullptr,
        ov_cursor.data_ptr<int32_t>(), ov_packed.numel(),
        (uint64_t*)tkeys.data_ptr<int64_t>(),
        (unsigned long long*)tvals.data_ptr<int64_t>(), mask,
        (int)region_bits, error_flag.data_ptr<int32_t>());
  };
  if (mode == AGG_COUNT) ov(k_overflow_agg<AGG_COUNT>);
  else ov(k_overflow_agg<AGG_SUM>);
}

// Split radix insert for the per-step Python engine's pipelined
// path: `radix_scatter_only` launches the (memset + per-segment
// scatter) on the CURRENT stream — the caller runs it under a side
// torch.cuda.Stream — and `radix_agg_only` drains a buffer set into
// the table on whatever stream is current.  Stream/event
// orchestration (scatter N+1 overlapping agg N, buffer-parity reuse
// fences, NCCL completion ordering) lives in Python with torch
// events, which keeps it per-state and lets the exchange wait be
// recorded on the side stream only.
void radix_scatter_only(
    torch::Tensor keys,
    torch::Tensor ts,
    c10::optional<torch::Tensor> vals,
    torch::Tensor max_ts,
    torch::Tensor error_flag,
    torch::Tensor gcursors,
    torch::Tensor ev_packed,
    torch::Tensor ev_vals,
    torch::Tensor ov_cursor,
    torch::Tensor ov_packed,
    torch::Tensor ov_vals,
    int64_t nslots,
    int64_t align_ms,
    int64_t len_ms,
    int64_t mode,
    int64_t ts_base,
    int64_t region_bits,
    int64_t off_ms,
    std::vector<int64_t> seg_counts,
    std::vector<int64_t> seg_bases) {
  if (off_ms <= 0) off_ms = len_ms;
  TORCH_CHECK(off_ms <= len_ms, "window offset must be <= length");
  check_dev(keys, torch::kInt32, "keys");
  bool seg32 = !seg_counts.empty();
  bool ts32 = ts.scalar_type() == torch::kInt32;
  if (seg32) {
    TORCH_CHECK(seg_counts.size() == seg_bases.size(),
                "seg_counts/seg_bases length mismatch");
    check_dev(ts, torch::kInt32, "ts32");
  } else if (!ts32) {
    check_dev(ts, torch::kInt64, "ts");
  }
  int64_t n = keys.numel();
  TORCH_CHECK((nslots & (nslots - 1)) == 0, "table size must be 2^k");
  int64_t nb = nslots >> region_bits;
  const int64_t* vptr = nullptr;
  if (mode == AGG_SUM) {
    TORCH_CHECK(vals.has_value(), "sum mode requires vals");
    vptr = vals->data_ptr<int64_t>();
  }
  auto stream = at::hip::getCurrentHIPStream();
  uint64_t mask = (uint64_t)(nslots - 1);
  dim3 block(256);
  uint64_t win_m2, win_maxfast;
  magic_div_u64(off_ms <= 0 ? len_ms : off_ms, &win_m2, &win_maxfast);

  // Same variant/segmentation decision as radix_agg_only (both read
  // the env once per call; one process = one configuration).
  ScatterKind kind = scatter_kind_env();
  int64_t xf = off_ms < len_ms ? (len_ms + off_ms - 1) / off_ms : 1;
  TORCH_CHECK(xf <= 64, "sliding expansion > 64x; use a larger offset");
  if (kind == SCAT_DIRECT && xf > 1) kind = SCAT_FIXED;
  int coarse = scatter_coarse_bits(kind);
  int seg_bits = (int)region_bits + coarse;
  int64_t nseg = nslots >> seg_bits;
  size_t staged_lds = (size_t)nseg * SC_GRAN * 8 *
                          (mode == AGG_SUM ? 2 : 1) +
                      4 * (size_t)nseg * sizeof(int);
  while (kind == SCAT_STAGED &&
         (nseg < 1 || staged_lds > 160 * 1024 || seg_bits > 13)) {
    if (coarse > 0 && seg_bits > 13) {
      coarse -= 1;
    } else {
      kind = SCAT_FIXED;
      coarse = 0;
    }
    seg_bits = (int)region_bits + coarse;
    nseg = nslots >> seg_bits;
    staged_lds = (size_t)nseg * SC_GRAN * 8 * (mode == AGG_SUM ? 2 : 1) +
                 4 * (size_t)nseg * sizeof(int);
  }
  if (seg_bits > 13) {
    coarse = 0;
    seg_bits = (int)region_bits;
    nseg = nb;
  }
  int64_t cap = ev_packed.numel() / nseg;
  if (kind == SCAT_STAGED) {
    cap &= ~(int64_t)(SC_GRAN - 1);
    if (cap < SC_GRAN) {
      kind = SCAT_FIXED;
      coarse = 0;
      seg_bits = (int)region_bits;
      nseg = nb;
      cap = ev_packed.numel() / nseg;
    }
  }
  TORCH_CHECK(cap * nseg >= 2 * n * xf || cap >= n * xf,
              "scatter buffers too small");
  size_t hist_lds = (size_t)nseg * sizeof(int);
  HIP_CHECK(hipMemsetAsync(
      gcursors.data_ptr<int32_t>(), 0, (size_t)nseg * sizeof(int), stream));
  HIP_CHECK(hipMemsetAsync(
      ov_cursor.data_ptr<int32_t>(), 0, sizeof(int), stream));
  if (n == 0) return;

  struct Seg {
    int64_t off, n, base;
  };
  std::vector<Seg> segs;
  if (seg32) {
    int64_t off = 0;
    for (size_t i = 0; i < seg_counts.size(); ++i) {
      if (seg_counts[i] > 0) segs.push_back({off, seg_counts[i], seg_bases[i]});
      off += seg_counts[i];
    }
  } else {
    segs.push_back({0, n, ts_base});
  }
  unsigned scat_threads = 256;
  auto scat = [&](auto kern, auto tsptr, const Seg& sg, unsigned gx,
                  size_t lds) {
    hipLaunchKernelGGL(
        kern, dim3(gx), dim3(scat_threads), lds,
        stream, keys.data_ptr<int32_t>() + sg.off, tsptr + sg.off,
        vptr != nullptr ? vptr + sg.off : nullptr, sg.n, align_ms, len_ms,
        off_ms, sg.base, mask, seg_bits, cap, gcursors.data_ptr<int32_t>(),
        (uint64_t*)ev_packed.data_ptr<int64_t>(),
        mode == AGG_SUM ? ev_vals.data_ptr<int64_t>() : nullptr,
        ov_cursor.data_ptr<int32_t>(),
        (uint64_t*)ov_packed.data_ptr<int64_t>(),
        mode == AGG_SUM ? ov_vals.data_ptr<int64_t>() : nullptr,
        ov_packed.numel(),
        (unsigned long long*)max_ts.data_ptr<int64_t>(),
        error_flag.data_ptr<int32_t>(), win_m2, win_maxfast);
  };
  auto scat_any = [&](auto tsptr, const Seg& sg) {
    using TSV = std::remove_const_t<std::remove_pointer_t<decltype(tsptr)>>;
    if (kind == SCAT_STAGED) {
      if (mode == AGG_COUNT) {
        // 512-thread workgroups (see radix_window_insert).
        unsigned gs = (unsigned)((sg.n + 8191) / 8192);
        if (gs > 1024) gs = 1024;
        if (gs < 1) gs = 1;
        scat_threads = 512;
        scat((k_radix_scatter_staged<AGG_COUNT, TSV, 16, 512>), tsptr,
             sg, gs, staged_lds);
        scat_threads = 256;
      } else {
        unsigned gs = (unsigned)((sg.n + 4095) / 4096);
        if (gs > 1024) gs = 1024;
        if (gs < 1) gs = 1;
        scat(k_radix_scatter_staged<AGG_SUM, TSV>, tsptr, sg, gs,
             staged_lds);
      }
    } else if (kind == SCAT_DIRECT) {
      unsigned gx = (unsigned)n_blocks(sg.n, 256);
      if (mode == AGG_COUNT)
        scat(k_radix_scatter_direct<AGG_COUNT, TSV>, tsptr, sg, gx, 0);
      else
        scat(k_radix_scatter_direct<AGG_SUM, TSV>, tsptr, sg, gx, 0);
    } else {
      unsigned gx = (unsigned)n_blocks(sg.n, 256);
      if (mode == AGG_COUNT)
        scat(k_radix_scatter_fixed<AGG_COUNT, TSV>, tsptr, sg, gx,
             2 * hist_lds);
      else
        scat(k_radix_scatter_fixed<AGG_SUM, TSV>, tsptr, sg, gx,
             2 * hist_lds);
    }
  };
  for (const Seg& sg : segs) {
    if (seg32 || ts32) scat_any(ts.data_ptr<int32_t>(), sg);
    else scat_any(ts.data_ptr<int64_t>(), sg);
  }
}

void radix_agg_only(
    torch::Tensor tkeys,
    torch::Tensor tvals,
    torch::Tensor error_flag,
    torch::Tensor gcursors,
    torch::Tensor ev_packed,
    torch::Tensor ev_vals,
    torch::Tensor ov_cursor,
    torch::Tensor ov_packed,
    torch::Tensor ov_vals,
    int64_t mode,
    int64_t region_bits) {
  int64_t nslots = tkeys.numel();
  int64_t nb = nslots >> region_bits;
  auto stream = at::hip::getCurrentHIPStream();
  uint64_t mask = (uint64_t)(nslots - 1);
  // Mirror radix_scatter_only's segmentation decision.
  ScatterKind kind = scatter_kind_env();
  int coarse = scatter_coarse_bits(kind);
  int seg_bits = (int)region_bits + coarse;
  int64_t nseg = nslots >> seg_bits;
  size_t staged_lds = (size_t)nseg * SC_GRAN * 8 *
                          (mode == AGG_SUM ? 2 : 1) +
                      4 * (size_t)nseg * sizeof(int);
  while (kind == SCAT_STAGED &&
         (nseg < 1 || staged_lds > 160 * 1024 || seg_bits > 13)) {
    if (coarse > 0 && seg_bits > 13) {
      coarse -= 1;
    } else {
      kind = SCAT_FIXED;
      coarse = 0;
    }
    seg_bits = (int)region_bits + coarse;
    nseg = nslots >> seg_bits;
    staged_lds = (size_t)nseg * SC_GRAN * 8 * (mode == AGG_SUM ? 2 : 1) +
                 4 * (size_t)nseg * sizeof(int);
  }
  if (seg_bits > 13) {
    seg_bits = (int)region_bits;
    nseg = nb;
  }
  int64_t cap = ev_packed.numel() / nseg;
  if (kind == SCAT_STAGED) {
    cap &= ~(int64_t)(SC_GRAN - 1);
    if (cap < SC_GRAN) {
      kind = SCAT_FIXED;
      seg_bits = (int)region_bits;
      nseg = nb;
      cap = ev_packed.numel() / nseg;
    }
  }
  auto offsets = at::arange(
      nseg, at::TensorOptions().dtype(at::kInt).device(tkeys.device()));
  offsets = offsets * (int)cap;
  size_t agg_lds = (size_t)16 << seg_bits;
  dim3 agg_block(seg_bits >= 12 ? 1024 : 256);
  auto agg = [&](auto kern) {
    hipLaunchKernelGGL(
        kern, dim3((unsigned)nseg), agg_block, agg_lds, stream,
        (const uint64_t*)ev_packed.data_ptr<int64_t>(),
        mode == AGG_SUM ? ev_vals.data_ptr<int64_t>() : nullptr,
        offsets.data_ptr<int32_t>(), gcursors.data_ptr<int32_t>(), cap,
        (uint64_t*)tkeys.data_ptr<int64_t>(),
        (unsigned long long*)tvals.data_ptr<int64_t>(), mask,
        (int)region_bits, seg_bits, error_flag.data_ptr<int32_t>());
  };
  if (mode == AGG_COUNT) agg(k_radix_agg<AGG_COUNT>);
  else agg(k_radix_agg<AGG_SUM>);
  auto ov = [&](auto kern) {
    hipLaunchKernelGGL(
        kern, dim3(64), dim3(256), 0, stream,
        (const uint64_t*)ov_packed.data_ptr<int64_t>(),
        mode == AGG_SUM ? ov_vals.data_ptr<int64_t>() : nullptr,
        ov_cursor.data_ptr<int32_t>(), ov_packed.numel(),
        (uint64_t*)tkeys.data_ptr<int64_t>(),
        (unsigned long long*)tvals.data_ptr<int64_t>(), mask,
        (int)region_bits, error_flag.data_ptr<int32_t>());
  };
  if (mode == AGG_COUNT) ov(k_overflow_agg<AGG_COUNT>);
  else ov(k_overflow_agg<AGG_SUM>);
}

void radix_v2_window_insert(
    torch::Tensor keys,
    torch::Tensor ts,
    torch::Tensor tkeys,
    torch::Tensor tvals,
    torch::Tensor max_ts,
    torch::Tensor error_flag,
    torch::Tensor gcur,    // int32 [V2_COARSE]
    torch::Tensor gres,    // int32 [V2_COARSE]
    torch::Tensor ev,      // int64 [V2_COARSE * cap_c]
    torch::Tensor ev_res,  // int64 [V2_COARSE * res_cap]
    int64_t align_ms,
    int64_t len_ms,
    int64_t ts_base,
    int64_t region_bits) {
  check_dev(keys, torch::kInt32, "keys");
  check_dev(ts, torch::kInt64, "ts");
  int64_t n = keys.numel();
  int64_t nslots = tkeys.numel();
  TORCH_CHECK((nslots & (nslots - 1)) == 0, "table size must be 2^k");
  int64_t cap_c = ev.numel() / V2_COARSE;
  TORCH_CHECK(cap_c % 8 == 0, "coarse capacity must be 8-aligned");
  int64_t res_cap = ev_res.numel() / V2_COARSE;
  if (n == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  uint64_t mask = (uint64_t)(nslots - 1);
  gcur.zero_();
  gres.zero_();
  hipLaunchKernelGGL(
      k_scatter_coarse, dim3(256), dim3(512), 0, stream,
      keys.data_ptr<int32_t>(), ts.data_ptr<int64_t>(), n, align_ms,
      len_ms, ts_base, mask, (int)region_bits, cap_c, res_cap,
      gcur.data_ptr<int32_t>(), gres.data_ptr<int32_t>(),
      (uint64_t*)ev.data_ptr<int64_t>(),
      (uint64_t*)ev_res.data_ptr<int64_t>(),
      (unsigned long long*)max_ts.data_ptr<int64_t>(),
      error_flag.data_ptr<int32_t>());
  int slices = 8;
  hipLaunchKernelGGL(
      k_agg_cached, dim3(V2_COARSE * slices), dim3(256), 0, stream,
      (const uint64_t*)ev.data_ptr<int64_t>(),
      (const uint64_t*)ev_res.data_ptr<int64_t>(),
      gcur.data_ptr<int32_t>(), gres.data_ptr<int32_t>(), cap_c, res_cap,
      slices, (uint64_t*)tkeys.data_ptr<int64_t>(),
      (unsigned long long*)tvals.data_ptr<int64_t>(), mask,
      (int)region_bits, error_flag.data_ptr<int32_t>());
}

int64_t close_extract(
    torch::Tensor tkeys,
    torch::Tensor tvals,
    int64_t win_lo,
    int64_t win_hi,
    torch::Tensor out_keys,
    torch::Tensor out_wins,
    torch::Tensor out_vals,
    torch::Tensor out_n) {
  check_dev(tkeys, torch::kInt64, "tkeys");
  check_dev(tvals, torch::kInt64, "tvals");
  check_dev(out_keys, torch::kInt32, "out_keys");
  check_dev(out_wins, torch::kInt32, "out_wins");
  check_dev(out_vals, torch::kInt64, "out_vals");
  check_dev(out_n, torch::kInt32, "out_n");
  int64_t nslots = tkeys.numel();
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(nslots, 256));
  hipLaunchKernelGGL(
      k_close_extract, grid, block, 0, stream,
      (const uint64_t*)tkeys.data_ptr<int64_t>(),
      (const unsigned long long*)tvals.data_ptr<int64_t>(), nslots, win_lo,
      win_hi, out_keys.data_ptr<int32_t>(),
      out_wins.data_ptr<int32_t>(), out_vals.data_ptr<int64_t>(),
      out_n.data_ptr<int32_t>(), out_keys.numel());
  return 0;
}

void close_migrate(
    torch::Tensor tkeys,
    torch::Tensor tvals,
    torch::Tensor nkeys,
    torch::Tensor nvals,
    int64_t win_hi,
    int64_t region_bits,
    torch::Tensor out_keys,
    torch::Tensor out_wins,
    torch::Tensor out_vals,
    torch::Tensor out_n,
    torch::Tensor error_flag) {
  check_dev(tkeys, torch::kInt64, "tkeys");
  check_dev(nkeys, torch::kInt64, "nkeys");
  int64_t nslots = tkeys.numel();
  TORCH_CHECK(nkeys.numel() == nslots, "table size mismatch");
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(nslots, 256));
  hipLaunchKernelGGL(
      k_close_migrate, grid, block, 0, stream,
      (const uint64_t*)tkeys.data_ptr<int64_t>(),
      (const unsigned long long*)tvals.data_ptr<int64_t>(), nslots, win_hi,
      (uint64_t*)nkeys.data_ptr<int64_t>(),
      (unsigned long long*)nvals.data_ptr<int64_t>(),
      (uint64_t)(nslots - 1), (int)region_bits,
      out_keys.data_ptr<int32_t>(), out_wins.data_ptr<int32_t>(),
      out_vals.data_ptr<int64_t>(), out_n.data_ptr<int32_t>(),
      out_keys.numel(), error_flag.data_ptr<int32_t>());
}

void stats_insert(
    torch::Tensor keys,
    torch::Tensor ts,
    torch::Tensor vals,
    torch::Tensor tkeys,
    torch::Tensor tcnt,
    torch::Tensor tsum,
    torch::Tensor tmin,
    torch::Tensor tmax,
    torch::Tensor max_ts,
    torch::Tensor error_flag,
    int64_t align_ms,
    int64_t len_ms,
    int64_t ts_base) {
  check_dev(keys, torch::kInt32, "keys");
  check_dev(ts, torch::kInt64, "ts");
  check_dev(vals, torch::kInt64, "vals");
  check_dev(tkeys, torch::kInt64, "tkeys");
  int64_t n = keys.numel();
  int64_t nslots = tkeys.numel();
  TORCH_CHECK((nslots & (nslots - 1)) == 0, "table size must be 2^k");
  if (n == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(n, 256));
  hipLaunchKernelGGL(
      k_stats_insert, grid, block, 0, stream, keys.data_ptr<int32_t>(),
      ts.data_ptr<int64_t>(), vals.data_ptr<int64_t>(),
      n, (uint64_t*)tkeys.data_ptr<int64_t>(),
      (long long*)tcnt.data_ptr<int64_t>(),
      (long long*)tsum.data_ptr<int64_t>(),
      (long long*)tmin.data_ptr<int64_t>(),
      (long long*)tmax.data_ptr<int64_t>(), (uint64_t)(nslots - 1),
      align_ms, len_ms, ts_base,
      (unsigned long long*)max_ts.data_ptr<int64_t>(),
      error_flag.data_ptr<int32_t>());
}

void radix_stats_insert(
    torch::Tensor keys,
    torch::Tensor ts,
    torch::Tensor vals,
    torch::Tensor tkeys,
    torch::Tensor tcnt,
    torch::Tensor tsum,
    torch::Tensor tmin,
    torch::Tensor tmax,
    torch::Tensor max_ts,
    torch::Tensor error_flag,
    torch::Tensor gcursors,
    torch::Tensor ev_packed,
    torch::Tensor ev_vals,
    torch::Tensor ov_cursor,
    torch::Tensor ov_packed,
    torch::Tensor ov_vals,
    int64_t align_ms,
    int64_t len_ms,
    int64_t ts_base,
    int64_t region_bits) {
  check_dev(keys, torch::kInt32, "keys");
  bool ts32 = ts.scalar_type() == torch::kInt32;
  if (!ts32) check_dev(ts, torch::kInt64, "ts");
  check_dev(vals, torch::kInt64, "vals");
  int64_t n = keys.numel();
  int64_t nslots = tkeys.numel();
  TORCH_CHECK((nslots & (nslots - 1)) == 0, "table size must be 2^k");
  TORCH_CHECK(region_bits > 0 && region_bits <= 10,
              "stats radix needs 0 < region_bits <= 10 (40 B/slot LDS)");
  int64_t nb = nslots >> region_bits;
  TORCH_CHECK(nb >= 1 && nb <= 8192, "region count out of range");
  if (n == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  uint64_t mask = (uint64_t)(nslots - 1);
  dim3 block(256);
  dim3 grid(n_blocks(n, 256));

  // Same scatter-variant decision as the window path; the stats agg
  // stages 40 B/LDS slot, capping coarse segments at lds_bits 11.
  ScatterKind kind = scatter_kind_env();
  int coarse = scatter_coarse_bits(kind);
  int seg_bits = (int)region_bits + coarse;
  int64_t nseg = nslots >> seg_bits;
  size_t staged_lds = (size_t)nseg * SC_GRAN * 8 * 2 +
                      4 * (size_t)nseg * sizeof(int);
  while (kind == SCAT_STAGED &&
         (nseg < 1 || staged_lds > 160 * 1024 || seg_bits > 11)) {
    if (coarse > 0 && seg_bits > 11) {
      coarse -= 1;
    } else {
      kind = SCAT_FIXED;
      coarse = 0;
    }
    seg_bits = (int)region_bits + coarse;
    nseg = nslots >> seg_bits;
    staged_lds = (size_t)nseg * SC_GRAN * 8 * 2 +
                 4 * (size_t)nseg * sizeof(int);
  }
  if (seg_bits > 11) {
    coarse = 0;
    seg_bits = (int)region_bits;
    nseg = nb;
  }
  int64_t cap = ev_packed.numel() / nseg;
  if (kind == SCAT_STAGED) {
    cap &= ~(int64_t)(SC_GRAN - 1);
    if (cap < SC_GRAN) {
      kind = SCAT_FIXED;
      coarse = 0;
      seg_bits = (int)region_bits;
      nseg = nb;
      cap = ev_packed.numel() / nseg;
    }
  }
  TORCH_CHECK(cap * nseg >= 2 * n || cap >= n,
              "scatter buffers too small (need ~2x batch)");
  TORCH_CHECK(ev_vals.numel() >= nseg * cap, "ev_vals too small");
  gcursors.narrow(0, 0, nseg).zero_();
  ov_cursor.zero_();
  size_t hist_lds = (size_t)nseg * sizeof(int);
  uint64_t win_m2, win_maxfast;
  magic_div_u64(len_ms, &win_m2, &win_maxfast);
  auto scat_any = [&](auto tsptr) {
    using TSV = std::remove_const_t<std::remove_pointer_t<decltype(tsptr)>>;
    if (kind == SCAT_STAGED) {
      unsigned gs = (unsigned)((n + 8191) / 8192);
      if (gs > 1024) gs = 1024;
      if (gs < 1) gs = 1;
      hipLaunchKernelGGL(
          (k_radix_scatter_staged<AGG_SUM, TSV, 16, 512>), dim3(gs),
          dim3(512), staged_lds,
          stream, keys.data_ptr<int32_t>(), tsptr,
          vals.data_ptr<int64_t>(), n, align_ms, len_ms, len_ms, ts_base,
          mask,
          seg_bits, cap, gcursors.data_ptr<int32_t>(),
          (uint64_t*)ev_packed.data_ptr<int64_t>(),
          ev_vals.data_ptr<int64_t>(), ov_cursor.data_ptr<int32_t>(),
          (uint64_t*)ov_packed.data_ptr<int64_t>(),
          ov_vals.data_ptr<int64_t>(), ov_packed.numel(),
          (unsigned long long*)max_ts.data_ptr<int64_t>(),
          error_flag.data_ptr<int32_t>(), win_m2, win_maxfast);
    } else {
      hipLaunchKernelGGL(
          (k_radix_scatter_fixed<AGG_SUM, TSV>), grid, block,
          2 * hist_lds, stream,
          keys.data_ptr<int32_t>(), tsptr,
          vals.data_ptr<int64_t>(), n, align_ms, len_ms, len_ms, ts_base,
          mask,
          seg_bits, cap, gcursors.data_ptr<int32_t>(),
          (uint64_t*)ev_packed.data_ptr<int64_t>(), ev_vals.data_ptr<int64_t>(),
          ov_cursor.data_ptr<int32_t>(),
          (uint64_t*)ov_packed.data_ptr<int64_t>(), ov_vals.data_ptr<int64_t>(),
          ov_packed.numel(),
          (unsigned long long*)max_ts.data_ptr<int64_t>(),
          error_flag.data_ptr<int32_t>(), win_m2, win_maxfast);
    }
  };
  if (ts32) scat_any(ts.data_ptr<int32_t>());
  else scat_any(ts.data_ptr<int64_t>());

  auto offsets = at::arange(
      nseg, at::TensorOptions().dtype(at::kInt).device(keys.device()));
  offsets = offsets * (int)cap;
  size_t agg_lds = (size_t)40 << seg_bits;
  dim3 agg_block(seg_bits >= 11 ? 1024 : 256);
  hipLaunchKernelGGL(
      k_radix_agg_stats, dim3((unsigned)nseg), agg_block, agg_lds, stream,
      (const uint64_t*)ev_packed.data_ptr<int64_t>(),
      ev_vals.data_ptr<int64_t>(), offsets.data_ptr<int32_t>(),
      gcursors.data_ptr<int32_t>(), (uint64_t*)tkeys.data_ptr<int64_t>(),
      (long long*)tcnt.data_ptr<int64_t>(),
      (long long*)tsum.data_ptr<int64_t>(),
      (long long*)tmin.data_ptr<int64_t>(),
      (long long*)tmax.data_ptr<int64_t>(), cap, mask, (int)region_bits,
      seg_bits, error_flag.data_ptr<int32_t>());

  hipLaunchKernelGGL(
      k_overflow_agg_stats, dim3(64), block, 0, stream,
      (const uint64_t*)ov_packed.data_ptr<int64_t>(),
      ov_vals.data_ptr<int64_t>(), ov_cursor.data_ptr<int32_t>(),
      ov_packed.numel(), (uint64_t*)tkeys.data_ptr<int64_t>(),
      (long long*)tcnt.data_ptr<int64_t>(),
      (long long*)tsum.data_ptr<int64_t>(),
      (long long*)tmin.data_ptr<int64_t>(),
      (long long*)tmax.data_ptr<int64_t>(), mask,
      error_flag.data_ptr<int32_t>());
}

void stats_fixup(
    torch::Tensor keys,
    torch::Tensor ts,
    torch::Tensor cnt_delta,
    torch::Tensor sum_delta,
    torch::Tensor tkeys,
    torch::Tensor tcnt,
    torch::Tensor tsum,
    int64_t align_ms,
    int64_t len_ms) {
  int64_t n = keys.numel();
  if (n == 0) return;
  int64_t nslots = tkeys.numel();
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(n, 256));
  hipLaunchKernelGGL(
      k_stats_fixup, grid, block, 0, stream, keys.data_ptr<int32_t>(),
      ts.data_ptr<int64_t>(), cnt_delta.data_ptr<int64_t>(),
      sum_delta.data_ptr<int64_t>(), n,
      (uint64_t*)tkeys.data_ptr<int64_t>(),
      (long long*)tcnt.data_ptr<int64_t>(),
      (long long*)tsum.data_ptr<int64_t>(), (uint64_t)(nslots - 1),
      align_ms, len_ms);
}

void stats_extract(
    torch::Tensor tkeys,
    torch::Tensor tcnt,
    torch::Tensor tsum,
    torch::Tensor tmin,
    torch::Tensor tmax,
    int64_t win_lo,
    int64_t win_hi,
    torch::Tensor out_keys,
    torch::Tensor out_wins,
    torch::Tensor out_cnt,
    torch::Tensor out_sum,
    torch::Tensor out_min,
    torch::Tensor out_max,
    torch::Tensor out_n) {
  int64_t nslots = tkeys.numel();
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(nslots, 256));
  hipLaunchKernelGGL(
      k_stats_extract, grid, block, 0, stream,
      (const uint64_t*)tkeys.data_ptr<int64_t>(),
      (const long long*)tcnt.data_ptr<int64_t>(),
      (const long long*)tsum.data_ptr<int64_t>(),
      (const long long*)tmin.data_ptr<int64_t>(),
      (const long long*)tmax.data_ptr<int64_t>(), nslots, win_lo,
      win_hi, out_keys.data_ptr<int32_t>(),
      out_wins.data_ptr<int32_t>(), out_cnt.data_ptr<int64_t>(),
      out_sum.data_ptr<int64_t>(), out_min.data_ptr<int64_t>(),
      out_max.data_ptr<int64_t>(), out_n.data_ptr<int32_t>(),
      out_keys.numel());
}

void join_insert(
    torch::Tensor keys,
    torch::Tensor vals,
    int64_t side,
    int64_t n_sides,
    torch::Tensor tkeys,
    torch::Tensor tval0,
    torch::Tensor tval1,
    torch::Tensor tflags,
    torch::Tensor out_keys,
    torch::Tensor out_v0,
    torch::Tensor out_v1,
    torch::Tensor out_n,
    torch::Tensor error_flag) {
  check_dev(keys, torch::kInt32, "keys");
  check_dev(vals, torch::kInt64, "vals");
  check_dev(tflags, torch::kInt32, "tflags");
  int64_t n = keys.numel();
  int64_t nslots = tkeys.numel();
  TORCH_CHECK((nslots & (nslots - 1)) == 0, "table size must be 2^k");
  TORCH_CHECK(n_sides == 2, "device join currently supports 2 sides");
  if (n == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(n, 256));
  hipLaunchKernelGGL(
      k_join_insert, grid, block, 0, stream, keys.data_ptr<int32_t>(),
      vals.data_ptr<int64_t>(), n, (int)side, (int)n_sides,
      (uint64_t*)tkeys.data_ptr<int64_t>(),
      (long long*)tval0.data_ptr<int64_t>(),
      (long long*)tval1.data_ptr<int64_t>(), tflags.data_ptr<int32_t>(),
      (uint64_t)(nslots - 1), out_keys.data_ptr<int32_t>(),
      out_v0.data_ptr<int64_t>(), out_v1.data_ptr<int64_t>(),
      out_n.data_ptr<int32_t>(), out_keys.numel(),
      error_flag.data_ptr<int32_t>());
}

void radix_join_insert(
    torch::Tensor keys,
    torch::Tensor zeros_ts,  // int64 [>= n] of zeros (key-only packing)
    torch::Tensor vals,
    int64_t side,
    int64_t n_sides,
    torch::Tensor tkeys,
    torch::Tensor tval0,
    torch::Tensor tval1,
    torch::Tensor tflags,
    torch::Tensor gcursors,
    torch::Tensor ev_packed,
    torch::Tensor ev_vals,
    torch::Tensor ov_cursor,
    torch::Tensor ov_packed,
    torch::Tensor ov_vals,
    torch::Tensor out_keys,
    torch::Tensor out_v0,
    torch::Tensor out_v1,
    torch::Tensor out_n,
    torch::Tensor max_ts_scratch,
    torch::Tensor error_flag,
    int64_t region_bits) {
  check_dev(keys, torch::kInt32, "keys");
  check_dev(vals, torch::kInt64, "vals");
  int64_t n = keys.numel();
  int64_t nslots = tkeys.numel();
  TORCH_CHECK((nslots & (nslots - 1)) == 0, "table size must be 2^k");
  TORCH_CHECK(n_sides == 2, "device join currently supports 2 sides");
  TORCH_CHECK(region_bits > 0 && region_bits <= 12, "bad region_bits");
  // Scatter segments one bit coarser than table regions
  // (BYTEWAX_JOIN_COARSE): halves the scatter's cursor LDS (better
  // occupancy, longer runs); the dedup/merge kernel probes the
  // global table by key, so segments need not match regions.
  int64_t coarse = 1;
  if (const char* c = std::getenv("BYTEWAX_JOIN_COARSE")) {
    if (c[0] >= '0' && c[0] <= '3' && c[1] == '\0') coarse = c[0] - '0';
  }
  int64_t seg_bits = region_bits + coarse;
  int64_t nb = nslots >> seg_bits;
  if (nb < 1) {
    nb = 1;
    seg_bits = 0;
    while (((int64_t)1 << seg_bits) < nslots) ++seg_bits;
  }
  int64_t cap = ev_packed.numel() / nb;
  // Staged (line-granule) scatter when the per-region capacity
  // supports it; the fixed variant otherwise.
  ScatterKind kind = scatter_kind_env() == SCAT_FIXED ? SCAT_FIXED
                                                      : SCAT_STAGED;
  size_t staged_lds =
      (size_t)nb * SC_GRAN * 8 * 2 + 4 * (size_t)nb * sizeof(int);
  if (kind == SCAT_STAGED) {
    cap &= ~(int64_t)(SC_GRAN - 1);
    if (cap < SC_GRAN || staged_lds > 160 * 1024) {
      kind = SCAT_FIXED;
      cap = ev_packed.numel() / nb;
    }
  }
  TORCH_CHECK(cap * nb >= 2 * n || cap >= n, "scatter buffers too small");
  if (n == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  uint64_t mask = (uint64_t)(nslots - 1);
  dim3 block(256);
  dim3 grid(n_blocks(n, 256));
  gcursors.narrow(0, 0, nb).zero_();
  ov_cursor.zero_();
  size_t hist_lds = (size_t)nb * sizeof(int);
  // Key-only packing: ts == 0, align 0, huge window -> win = 0.
  // 512-thread workgroups + grid 1024: same tuning as the stats SUM
  // scatter (profiles/r02 call 29: the scatter is 63% of join time).
  if (kind == SCAT_STAGED) {
    unsigned gs = (unsigned)((n + 8191) / 8192);
    if (gs > 1024) gs = 1024;
    if (gs < 1) gs = 1;
    hipLaunchKernelGGL(
        (k_radix_scatter_staged<AGG_SUM, int64_t, 16, 512>), dim3(gs),
        dim3(512), staged_lds,
        stream, keys.data_ptr<int32_t>(), zeros_ts.data_ptr<int64_t>(),
        vals.data_ptr<int64_t>(), n, 0, (int64_t)1 << 40,
        (int64_t)1 << 40, 0, mask,
        (int)seg_bits, cap, gcursors.data_ptr<int32_t>(),
        (uint64_t*)ev_packed.data_ptr<int64_t>(), ev_vals.data_ptr<int64_t>(),
        ov_cursor.data_ptr<int32_t>(),
        (uint64_t*)ov_packed.data_ptr<int64_t>(), ov_vals.data_ptr<int64_t>(),
        ov_packed.numel(),
        (unsigned long long*)max_ts_scratch.data_ptr<int64_t>(),
        error_flag.data_ptr<int32_t>(), (uint64_t)1 << 24, ~(uint64_t)0);
  } else {
    hipLaunchKernelGGL(
        k_radix_scatter_fixed<AGG_SUM>, grid, block, 2 * hist_lds, stream,
        keys.data_ptr<int32_t>(), zeros_ts.data_ptr<int64_t>(),
        vals.data_ptr<int64_t>(), n, 0, (int64_t)1 << 40, (int64_t)1 << 40,
        0, mask,
        (int)seg_bits, cap, gcursors.data_ptr<int32_t>(),
        (uint64_t*)ev_packed.data_ptr<int64_t>(), ev_vals.data_ptr<int64_t>(),
        ov_cursor.data_ptr<int32_t>(),
        (uint64_t*)ov_packed.data_ptr<int64_t>(), ov_vals.data_ptr<int64_t>(),
        ov_packed.numel(),
        (unsigned long long*)max_ts_scratch.data_ptr<int64_t>(),
        error_flag.data_ptr<int32_t>(), (uint64_t)1 << 24, ~(uint64_t)0);
  }
  // LDS staging sized one bit above the segment's table span
  // (2^seg_bits cells per segment); headroom keeps the in-LDS probe
  // chains short at high per-segment cardinality.
  int lds_bits = (int)seg_bits + 1;
  if (lds_bits < 6) lds_bits = 6;
  if (lds_bits > 12) lds_bits = 12;
  // 8 B key + 8 B value + 4 B event count per LDS slot.
  size_t join_lds = (size_t)20 << lds_bits;
  if (join_lds_env() && join_lds <= 144 * 1024) {
    // 1024 threads/workgroup: the merge kernel is LDS-atomic latency
    // bound, so more waves in flight per block hide it (sweep r02
    // call 31: 256 -> 17.3, 512 -> 20.2, 1024 -> 20.5 Ge/s;
    // BYTEWAX_JOIN_THREADS overrides).
    int jthreads = 1024;
    if (const char* t = std::getenv("BYTEWAX_JOIN_THREADS")) {
      int v = std::atoi(t);
      if (v == 256 || v == 512 || v == 1024) jthreads = v;
    }
    auto launch_lds = [&](auto kern, int thr) {
      hipLaunchKernelGGL(
          kern, dim3((unsigned)nb), dim3(thr), join_lds, stream,
          (const uint64_t*)ev_packed.data_ptr<int64_t>(),
          ev_vals.data_ptr<int64_t>(), gcursors.data_ptr<int32_t>(), cap,
          (int)side, (int)n_sides, (uint64_t*)tkeys.data_ptr<int64_t>(),
          (long long*)tval0.data_ptr<int64_t>(),
          (long long*)tval1.data_ptr<int64_t>(), tflags.data_ptr<int32_t>(),
          mask, (int)region_bits, lds_bits, out_keys.data_ptr<int32_t>(),
          out_v0.data_ptr<int64_t>(), out_v1.data_ptr<int64_t>(),
          out_n.data_ptr<int32_t>(), out_keys.numel(),
          error_flag.data_ptr<int32_t>());
    };
    if (jthreads == 1024) launch_lds(k_join_region_lds<1024>, 1024);
    else if (jthreads == 512) launch_lds(k_join_region_lds<512>, 512);
    else launch_lds(k_join_region_lds<256>, 256);
  } else {
    hipLaunchKernelGGL(
        k_join_region, dim3((unsigned)nb), block, 0, stream,
        (const uint64_t*)ev_packed.data_ptr<int64_t>(),
        ev_vals.data_ptr<int64_t>(), gcursors.data_ptr<int32_t>(), cap,
        (int)side, (int)n_sides, (uint64_t*)tkeys.data_ptr<int64_t>(),
        (long long*)tval0.data_ptr<int64_t>(),
        (long long*)tval1.data_ptr<int64_t>(), tflags.data_ptr<int32_t>(),
        mask, (int)region_bits, out_keys.data_ptr<int32_t>(),
        out_v0.data_ptr<int64_t>(), out_v1.data_ptr<int64_t>(),
        out_n.data_ptr<int32_t>(), out_keys.numel(),
        error_flag.data_ptr<int32_t>());
  }
  hipLaunchKernelGGL(
      k_join_overflow, dim3(64), block, 0, stream,
      (const uint64_t*)ov_packed.data_ptr<int64_t>(),
      ov_vals.data_ptr<int64_t>(), ov_cursor.data_ptr<int32_t>(),
      ov_packed.numel(), (int)side, (int)n_sides,
      (uint64_t*)tkeys.data_ptr<int64_t>(),
      (long long*)tval0.data_ptr<int64_t>(),
      (long long*)tval1.data_ptr<int64_t>(), tflags.data_ptr<int32_t>(),
      mask, (int)region_bits, out_keys.data_ptr<int32_t>(),
      out_v0.data_ptr<int64_t>(), out_v1.data_ptr<int64_t>(),
      out_n.data_ptr<int32_t>(), out_keys.numel(),
      error_flag.data_ptr<int32_t>());
}

// Batch-level session merge: when the gap is at least the batch's
// time span, no session boundary can fall INSIDE the batch, so each
// key's per-batch (count, min_ts, max_ts) — computed by the radix
// stats kernels at full speed — is merged as one unit.  Requires
// watermark-ordered batches (each batch's events at/after the
// previous batch's), the same contract as the sorted walk.
__global__ void k_session_merge_batch(
    const int32_t* __restrict__ keys,
    const int64_t* __restrict__ cnt,
    const int64_t* __restrict__ mn_ts,
    const int64_t* __restrict__ mx_ts,
    const int* __restrict__ n_rows,
    int64_t ts_shift,  // add to mn/mx (zero-based batch timestamps)
    int64_t gap_ms,
    uint64_t* __restrict__ skeys,
    long long* __restrict__ sstart,
    long long* __restrict__ slast,
    long long* __restrict__ sacc,
    uint64_t mask,
    int32_t* __restrict__ out_keys,
    int64_t* __restrict__ out_start,
    int64_t* __restrict__ out_end,
    int64_t* __restrict__ out_vals,
    int* __restrict__ out_n,
    int64_t out_cap,
    unsigned long long* __restrict__ max_ts,
    int* __restrict__ error_flag) {
  int64_t n = *n_rows;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  int64_t local_max = 0;
  for (; i < n; i += stride) {
    uint64_t key = (uint64_t)(uint32_t)keys[i];
    uint64_t slot = session_find_or_claim(skeys, mask, key);
    if (slot == ~0ULL) {
      atomicExch(error_flag, 1);
      continue;
    }
    long long lo = mn_ts[i] + ts_shift;
    long long hi = mx_ts[i] + ts_shift;
    long long c = cnt[i];
    if (hi > local_max) local_max = hi;
    long long last = slast[slot];
    if (last >= 0 && lo - last > gap_ms) {
      int idx = atomicAdd(out_n, 1);
      if (idx < out_cap) {
        out_keys[idx] = (int32_t)(uint32_t)key;
        out_start[idx] = sstart[slot];
        out_end[idx] = last;
        out_vals[idx] = sacc[slot];
      } else {
        atomicExch(error_flag, 1);
      }
      last = -1;
    }
    if (last < 0) {
      sstart[slot] = lo;
      sacc[slot] = c;
    } else {
      sacc[slot] += c;
    }
    slast[slot] = hi;
  }
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    int64_t other = __shfl_down((long long)local_max, off);
    if (other > local_max) local_max = other;
  }
  if ((threadIdx.x & (WAVE - 1)) == 0 && local_max > 0) {
    atomicMax(max_ts, (unsigned long long)local_max);
  }
}

__global__ void k_session_restore(
    const int32_t* __restrict__ keys,
    const int64_t* __restrict__ start,
    const int64_t* __restrict__ last,
    const int64_t* __restrict__ acc,
    int64_t n,
    uint64_t* __restrict__ skeys,
    long long* __restrict__ sstart,
    long long* __restrict__ slast,
    long long* __restrict__ sacc,
    uint64_t mask,
    int* __restrict__ error_flag) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    uint64_t slot = session_find_or_claim(
        skeys, mask, (uint64_t)(uint32_t)keys[i]);
    if (slot == ~0ULL) {
      atomicExch(error_flag, 1);
      continue;
    }
    sstart[slot] = start[i];
    slast[slot] = last[i];
    sacc[slot] = acc[i];
  }
}

// Fused radix session aggregation (the fast path's round-2 rework):
// an AGG_TS scatter partitions (key, t) pairs into per-segment runs;
// one block then LDS-aggregates its segment's per-key
// (count, min_ts, max_ts) and merges each distinct key ONCE into the
// session table with the k_session_merge_batch semantics.  Replaces
// the stats-table round trip (insert + extract + merge + 5 table
// clears per batch).  Contract: gap >= batch time span and
// watermark-ordered batches (same as the batch merge).  min/max
// guards keep multi-merge (overflow spill) within one batch exact.
__device__ inline void session_merge_cell(
    uint64_t* skeys, long long* sstart, long long* slast,
    long long* sacc, uint64_t mask, uint64_t key, long long lo,
    long long hi, long long c, int64_t gap_ms,
    int32_t* out_keys, int64_t* out_start, int64_t* out_end,
    int64_t* out_vals, int* out_n, int64_t out_cap, int* error_flag) {
  uint64_t slot = session_find_or_claim(skeys, mask, key);
  if (slot == ~0ULL) {
    atomicExch(error_flag, 1);
    return;
  }
  long long last = slast[slot];
  if (last >= 0 && lo - last > gap_ms) {
    int idx = atomicAdd(out_n, 1);
    if (idx < out_cap) {
      out_keys[idx] = (int32_t)(uint32_t)key;
      out_start[idx] = sstart[slot];
      out_end[idx] = last;
      out_vals[idx] = sacc[slot];
    } else {
      atomicExch(error_flag, 1);
    }
    last = -1;
  }
  if (last < 0) {
    sstart[slot] = lo;
    sacc[slot] = c;
    slast[slot] = hi;
  } else {
    sacc[slot] += c;
    if (lo < sstart[slot]) sstart[slot] = lo;
    if (hi > slast[slot]) slast[slot] = hi;
  }
}

__global__ __launch_bounds__(1024) void k_radix_session_agg(
    const uint64_t* __restrict__ ev_packed,  // keys (AGG_TS packing)
    const int64_t* __restrict__ ev_vals,     // absolute timestamps
    const int* __restrict__ offsets,
    const int* __restrict__ counts,
    int64_t clamp_cap,
    int lds_bits,
    uint64_t* __restrict__ skeys,
    long long* __restrict__ sstart,
    long long* __restrict__ slast,
    long long* __restrict__ sacc,
    uint64_t mask,
    int64_t gap_ms,
    int* __restrict__ ov_cursor,
    uint64_t* __restrict__ ov_packed,
    int64_t* __restrict__ ov_vals,
    int64_t ov_cap,
    int32_t* __restrict__ out_keys,
    int64_t* __restrict__ out_start,
    int64_t* __restrict__ out_end,
    int64_t* __restrict__ out_vals,
    int* __restrict__ out_n,
    int64_t out_cap,
    int* __restrict__ error_flag) {
  extern __shared__ char smem[];
  int slots = 1 << lds_bits;
  uint64_t* lkeys = (uint64_t*)smem;
  long long* lcnt = (long long*)(smem + (size_t)slots * 8);
  long long* lmn = lcnt + slots;
  long long* lmx = lmn + slots;
  for (int s = threadIdx.x; s < slots; s += blockDim.x) {
    lkeys[s] = EMPTY_SLOT;
    lcnt[s] = 0;
    lmn[s] = 0x7FFFFFFFFFFFFFFFLL;
    lmx[s] = -0x7FFFFFFFFFFFFFFFLL;
  }
  __syncthreads();
  int b = blockIdx.x;
  int cnt = counts[b];
  if (clamp_cap > 0 && cnt > (int)clamp_cap) cnt = (int)clamp_cap;
  int start = offsets[b];
  for (int j = threadIdx.x; j < cnt; j += blockDim.x) {
    uint64_t key = ev_packed[start + j];
    if (key == EMPTY_SLOT) continue;  // staged-scatter pad
    long long t = ev_vals[start + j];
    uint64_t h64 = mix64(key);
    int lh = (int)((h64 >> 32) & (slots - 1));
    bool done = false;
    for (int p = 0; p < slots; ++p) {
      uint64_t cur = lkeys[lh];
      if (cur == EMPTY_SLOT) {
        uint64_t prev = atomicCAS(
            (unsigned long long*)&lkeys[lh], EMPTY_SLOT, key);
        cur = (prev == EMPTY_SLOT) ? key : prev;
      }
      if (cur == key) {
        atomicAdd((unsigned long long*)&lcnt[lh], 1ULL);
        atomicMin(&lmn[lh], t);
        atomicMax(&lmx[lh], t);
        done = true;
        break;
      }
      lh = (lh + 1) & (slots - 1);
    }
    if (!done) {
      // LDS staging full: spill to the (sequential) overflow walk.
      int opos = atomicAdd(ov_cursor, 1);
      if (opos < ov_cap) {
        ov_packed[opos] = key;
        ov_vals[opos] = t;
      } else {
        atomicExch(error_flag, 1);
      }
    }
  }
  __syncthreads();
  for (int s = threadIdx.x; s < slots; s += blockDim.x) {
    if (lkeys[s] == EMPTY_SLOT) continue;
    session_merge_cell(
        skeys, sstart, slast, sacc, mask, lkeys[s], lmn[s], lmx[s],
        lcnt[s], gap_ms, out_keys, out_start, out_end, out_vals, out_n,
        out_cap, error_flag);
  }
}

// Sequential overflow walk: one thread merges spilled (key, t) pairs
// per event — order within the batch does not matter under the
// gap >= span contract (no close can trigger between same-batch
// merges; min/max guards keep start/last exact).  Slow by design;
// reached only under extreme key skew or LDS overflow.
__global__ void k_session_ov_walk(
    const uint64_t* __restrict__ ov_packed,
    const int64_t* __restrict__ ov_vals,
    const int* __restrict__ ov_cursor,
    int64_t ov_cap,
    uint64_t* __restrict__ skeys,
    long long* __restrict__ sstart,
    long long* __restrict__ slast,
    long long* __restrict__ sacc,
    uint64_t mask,
    int64_t gap_ms,
    int32_t* __restrict__ out_keys,
    int64_t* __restrict__ out_start,
    int64_t* __restrict__ out_end,
    int64_t* __restrict__ out_vals,
    int* __restrict__ out_n,
    int64_t out_cap,
    int* __restrict__ error_flag) {
  if (blockIdx.x != 0 || threadIdx.x != 0) return;
  int64_t n = *ov_cursor;
  if (n > ov_cap) n = ov_cap;
  for (int64_t i = 0; i < n; ++i) {
    session_merge_cell(
        skeys, sstart, slast, sacc, mask, ov_packed[i], ov_vals[i],
        ov_vals[i], 1, gap_ms, out_keys, out_start, out_end, out_vals,
        out_n, out_cap, error_flag);
  }
}

void session_insert(
    torch::Tensor keys,  // SORTED by (key, ts)
    torch::Tensor ts,
    c10::optional<torch::Tensor> vals,
    torch::Tensor seg_start,
    torch::Tensor seg_end,
    torch::Tensor skeys,
    torch::Tensor sstart,
    torch::Tensor slast,
    torch::Tensor sacc,
    torch::Tensor out_keys,
    torch::Tensor out_start,
    torch::Tensor out_end,
    torch::Tensor out_vals,
    torch::Tensor out_n,
    torch::Tensor max_ts,
    torch::Tensor error_flag,
    int64_t gap_ms,
    int64_t mode) {
  check_dev(keys, torch::kInt32, "keys");
  check_dev(ts, torch::kInt64, "ts");
  int64_t nslots = skeys.numel();
  TORCH_CHECK((nslots & (nslots - 1)) == 0, "table size must be 2^k");
  int64_t n_segs = seg_start.numel();
  if (n_segs == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(n_segs, 256));
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(
        kern, grid, block, 0, stream, keys.data_ptr<int32_t>(),
        ts.data_ptr<int64_t>(),
        mode == AGG_SUM ? vals->data_ptr<int64_t>() : nullptr,
        seg_start.data_ptr<int64_t>(), seg_end.data_ptr<int64_t>(),
        n_segs, gap_ms, (uint64_t*)skeys.data_ptr<int64_t>(),
        (long long*)sstart.data_ptr<int64_t>(),
        (long long*)slast.data_ptr<int64_t>(),
        (long long*)sacc.data_ptr<int64_t>(),
        (uint64_t)(nslots - 1), out_keys.data_ptr<int32_t>(),
        out_start.data_ptr<int64_t>(), out_end.data_ptr<int64_t>(),
        out_vals.data_ptr<int64_t>(), out_n.data_ptr<int32_t>(),
        out_keys.numel(),
        (unsigned long long*)max_ts.data_ptr<int64_t>(),
        error_flag.data_ptr<int32_t>());
  };
  if (mode == AGG_COUNT) launch(k_session_insert<AGG_COUNT>);
  else launch(k_session_insert<AGG_SUM>);
}

void session_close_migrate(
    torch::Tensor skeys,
    torch::Tensor sstart,
    torch::Tensor slast,
    torch::Tensor sacc,
    torch::Tensor dst_keys,
    torch::Tensor dst_start,
    torch::Tensor dst_last,
    torch::Tensor dst_acc,
    torch::Tensor out_keys,
    torch::Tensor out_start,
    torch::Tensor out_end,
    torch::Tensor out_vals,
    torch::Tensor out_n,
    torch::Tensor error_flag,
    int64_t horizon_ms) {
  int64_t nslots = skeys.numel();
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  hipLaunchKernelGGL(
      k_session_close_migrate, dim3(n_blocks(nslots, 256)), block, 0,
      stream, (const uint64_t*)skeys.data_ptr<int64_t>(),
      (const long long*)sstart.data_ptr<int64_t>(),
      (const long long*)slast.data_ptr<int64_t>(),
      (const long long*)sacc.data_ptr<int64_t>(), nslots, horizon_ms,
      (uint64_t*)dst_keys.data_ptr<int64_t>(),
      (long long*)dst_start.data_ptr<int64_t>(),
      (long long*)dst_last.data_ptr<int64_t>(),
      (long long*)dst_acc.data_ptr<int64_t>(),
      (uint64_t)(nslots - 1), out_keys.data_ptr<int32_t>(),
      out_start.data_ptr<int64_t>(), out_end.data_ptr<int64_t>(),
      out_vals.data_ptr<int64_t>(), out_n.data_ptr<int32_t>(),
      out_keys.numel(), error_flag.data_ptr<int32_t>());
}

void session_merge_batch(
    torch::Tensor keys,
    torch::Tensor cnt,
    torch::Tensor mn_ts,
    torch::Tensor mx_ts,
    torch::Tensor n_rows,
    torch::Tensor skeys,
    torch::Tensor sstart,
    torch::Tensor slast,
    torch::Tensor sacc,
    torch::Tensor out_keys,
    torch::Tensor out_start,
    torch::Tensor out_end,
    torch::Tensor out_vals,
    torch::Tensor out_n,
    torch::Tensor max_ts,
    torch::Tensor error_flag,
    int64_t ts_shift,
    int64_t gap_ms) {
  int64_t nslots = skeys.numel();
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  hipLaunchKernelGGL(
      k_session_merge_batch, dim3(n_blocks(out_keys.numel(), 256)),
      block, 0, stream, keys.data_ptr<int32_t>(),
      cnt.data_ptr<int64_t>(), mn_ts.data_ptr<int64_t>(),
      mx_ts.data_ptr<int64_t>(), n_rows.data_ptr<int32_t>(), ts_shift,
      gap_ms, (uint64_t*)skeys.data_ptr<int64_t>(),
      (long long*)sstart.data_ptr<int64_t>(),
      (long long*)slast.data_ptr<int64_t>(),
      (long long*)sacc.data_ptr<int64_t>(),
      (uint64_t)(nslots - 1), out_keys.data_ptr<int32_t>(),
      out_start.data_ptr<int64_t>(), out_end.data_ptr<int64_t>(),
      out_vals.data_ptr<int64_t>(), out_n.data_ptr<int32_t>(),
      out_keys.numel(),
      (unsigned long long*)max_ts.data_ptr<int64_t>(),
      error_flag.data_ptr<int32_t>());
}

void session_restore(
    torch::Tensor keys,
    torch::Tensor start,
    torch::Tensor last,
    torch::Tensor acc,
    torch::Tensor skeys,
    torch::Tensor sstart,
    torch::Tensor slast,
    torch::Tensor sacc,
    torch::Tensor error_flag) {
  check_dev(keys, torch::kInt32, "keys");
  int64_t n = keys.numel();
  if (n == 0) return;
  int64_t nslots = skeys.numel();
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  hipLaunchKernelGGL(
      k_session_restore, dim3(n_blocks(n, 256)), block, 0, stream,
      keys.data_ptr<int32_t>(), start.data_ptr<int64_t>(),
      last.data_ptr<int64_t>(), acc.data_ptr<int64_t>(), n,
      (uint64_t*)skeys.data_ptr<int64_t>(),
      (long long*)sstart.data_ptr<int64_t>(),
      (long long*)slast.data_ptr<int64_t>(),
      (long long*)sacc.data_ptr<int64_t>(),
      (uint64_t)(nslots - 1), error_flag.data_ptr<int32_t>());
}

void join_extract(
    torch::Tensor tkeys,
    torch::Tensor tval0,
    torch::Tensor tval1,
    torch::Tensor tflags,
    torch::Tensor out_keys,
    torch::Tensor out_v0,
    torch::Tensor out_v1,
    torch::Tensor out_flags,
    torch::Tensor out_n) {
  int64_t nslots = tkeys.numel();
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(nslots, 256));
  hipLaunchKernelGGL(
      k_join_extract, grid, block, 0, stream,
      (uint64_t*)tkeys.data_ptr<int64_t>(),
      (long long*)tval0.data_ptr<int64_t>(),
      (long long*)tval1.data_ptr<int64_t>(), tflags.data_ptr<int32_t>(),
      nslots, out_keys.data_ptr<int32_t>(), out_v0.data_ptr<int64_t>(),
      out_v1.data_ptr<int64_t>(), out_flags.data_ptr<int32_t>(),
      out_n.data_ptr<int32_t>(), out_keys.numel());
}

int64_t filter_compact(
    torch::Tensor keys,
    torch::Tensor ts,
    c10::optional<torch::Tensor> vals,
    torch::Tensor mask,
    torch::Tensor out_keys,
    torch::Tensor out_ts,
    torch::Tensor out_vals,
    torch::Tensor out_n) {
  check_dev(keys, torch::kInt32, "keys");
  check_dev(ts, torch::kInt64, "ts");
  TORCH_CHECK(mask.scalar_type() == torch::kUInt8 ||
                  mask.scalar_type() == torch::kBool,
              "mask must be bool/uint8");
  int64_t n = keys.numel();
  if (n == 0) return 0;
  const int64_t* vptr = nullptr;
  if (vals.has_value()) vptr = vals->data_ptr<int64_t>();
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(n, 256));
  hipLaunchKernelGGL(
      k_filter_compact, grid, block, 0, stream, keys.data_ptr<int32_t>(),
      ts.data_ptr<int64_t>(), vptr, (const uint8_t*)mask.data_ptr(), n,
      out_keys.data_ptr<int32_t>(), out_ts.data_ptr<int64_t>(),
      out_vals.data_ptr<int64_t>(), out_n.data_ptr<int32_t>());
  return 0;
}

void bucket_hist(torch::Tensor keys, int64_t world, torch::Tensor counts) {
  check_dev(keys, torch::kInt32, "keys");
  check_dev(counts, torch::kInt32, "counts");
  int64_t n = keys.numel();
  if (n == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(n, 256));
  hipLaunchKernelGGL(
      k_bucket_hist, grid, block, (int)(world * sizeof(int)), stream,
      keys.data_ptr<int32_t>(), n, (int)world, counts.data_ptr<int32_t>());
}

void bucket_scatter(
    torch::Tensor keys,
    torch::Tensor ts,
    c10::optional<torch::Tensor> vals,
    int64_t world,
    torch::Tensor cursors,
    torch::Tensor out_keys,
    torch::Tensor out_ts,
    torch::Tensor out_vals) {
  check_dev(keys, torch::kInt32, "keys");
  bool ts32 = ts.scalar_type() == torch::kInt32;
  if (!ts32) check_dev(ts, torch::kInt64, "ts");
  TORCH_CHECK(out_ts.scalar_type() == ts.scalar_type(),
              "out_ts dtype must match ts");
  check_dev(cursors, torch::kInt32, "cursors");
  int64_t n = keys.numel();
  if (n == 0) return;
  const int64_t* vptr = nullptr;
  if (vals.has_value()) {
    check_dev(*vals, torch::kInt64, "vals");
    vptr = vals->data_ptr<int64_t>();
  }
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(n, 256));
  if (ts32) {
    hipLaunchKernelGGL(
        k_bucket_scatter<int32_t>, grid, block, 0, stream,
        keys.data_ptr<int32_t>(), ts.data_ptr<int32_t>(), vptr, n,
        (int)world, cursors.data_ptr<int32_t>(),
        out_keys.data_ptr<int32_t>(), out_ts.data_ptr<int32_t>(),
        out_vals.data_ptr<int64_t>());
  } else {
    hipLaunchKernelGGL(
        k_bucket_scatter<int64_t>, grid, block, 0, stream,
        keys.data_ptr<int32_t>(), ts.data_ptr<int64_t>(), vptr, n,
        (int)world, cursors.data_ptr<int32_t>(),
        out_keys.data_ptr<int32_t>(), out_ts.data_ptr<int64_t>(),
        out_vals.data_ptr<int64_t>());
  }
}

// ---------------------------------------------------------------------------
// Native step-loop executor: runs the steady-state columnar window
// pipeline (source batch -> fused insert -> watermark close) for a
// whole run of steps with NO Python between steps.  This is the role
// the reference's Rust `step_or_park` worker loop plays
// (reference src/worker.rs:68-83), specialized for the device path:
// the host thread is just a kernel-launch engine on one HIP stream.
// Returns the total number of closed-window rows; per-step host
// launch timestamps (ns) are written into `step_ns_out` for latency
// percentiles.  The GIL is released for the duration.
// ---------------------------------------------------------------------------

int64_t native_run_window_steps(
    std::vector<torch::Tensor> key_pool,
    std::vector<torch::Tensor> ts_pool,
    int64_t start_step,
    int64_t n_steps,
    int64_t sim_ms_per_batch,
    torch::Tensor tkeys,
    torch::Tensor tvals,
    torch::Tensor max_ts,
    torch::Tensor error_flag,
    torch::Tensor out_keys,
    torch::Tensor out_wins,
    torch::Tensor out_vals,
    torch::Tensor out_n,
    int64_t align_ms,
    int64_t len_ms,
    int64_t wait_ms,
    int64_t mode,
    bool dedup,
    int64_t closed_horizon_in,
    torch::Tensor step_ns_out,  // int64 CPU tensor [n_steps]
    // int64 CPU tensor [3]: closed_horizon, rows, table-swap parity
    torch::Tensor state_out,
    int64_t region_bits,
    bool use_radix,
    c10::optional<torch::Tensor> gcursors,
    c10::optional<torch::Tensor> ev_packed,
    c10::optional<torch::Tensor> ev_vals,
    c10::optional<torch::Tensor> ov_cursor,
    c10::optional<torch::Tensor> ov_packed,
    c10::optional<torch::Tensor> ov_vals,
    torch::Tensor alt_tkeys,
    torch::Tensor alt_tvals,
    bool use_radix_v2,
    c10::optional<torch::Tensor> v2_gcur,
    c10::optional<torch::Tensor> v2_gres,
    c10::optional<torch::Tensor> v2_ev,
    c10::optional<torch::Tensor> v2_ev_res,
    bool pipelined = false,
    c10::optional<torch::Tensor> gcursors2 = c10::nullopt,
    c10::optional<torch::Tensor> ev_packed2 = c10::nullopt,
    c10::optional<torch::Tensor> ov_cursor2 = c10::nullopt,
    c10::optional<torch::Tensor> ov_packed2 = c10::nullopt) {
  TORCH_CHECK(!key_pool.empty(), "empty key pool");
  int64_t nslots = tkeys.numel();
  TORCH_CHECK((nslots & (nslots - 1)) == 0, "table size must be 2^k");
  int64_t n = key_pool[0].numel();
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(n, 256));
  int64_t* step_ns =
      step_ns_out.numel() >= n_steps ? step_ns_out.data_ptr<int64_t>()
                                     : nullptr;

  int64_t closed_horizon = closed_horizon_in;
  int64_t total_rows = 0;
  int pool = (int)key_pool.size();
  // Double-buffered tables: close = emit + migrate live cells into the
  // (pre-reset) alternate table, then swap and reset the old one.
  torch::Tensor cur_k = tkeys, cur_v = tvals;
  torch::Tensor alt_k = alt_tkeys, alt_v = alt_tvals;
  int swap_parity = 0;

  // Pinned host scalar for the close-count readback.
  int* h_n = nullptr;
  HIP_CHECK(hipHostMalloc((void**)&h_n, sizeof(int), hipHostMallocDefault));

  // Pipelined radix: the scatter of step N+1 runs on its own stream,
  // into the OTHER of two region-buffer sets, while the aggregation
  // of step N drains the first on the main stream — the ~0.4 ms agg
  // hides entirely behind the ~1.4 ms scatter.  Events serialize
  // scatter(N)->agg(N) and agg(N)->scatter(N+2) (buffer reuse).
  bool pipe = pipelined && use_radix && !use_radix_v2 &&
              gcursors2.has_value();
  hipStream_t sc_stream = nullptr;
  hipEvent_t ev_sc[2] = {nullptr, nullptr};
  hipEvent_t ev_ag[2] = {nullptr, nullptr};
  torch::Tensor agg_offsets;
  int64_t nb = 0, cap = 0, nseg = 0;
  int seg_bits = (int)region_bits;
  ScatterKind kind = SCAT_FIXED;
  size_t staged_lds = 0;
  uint64_t mask = (uint64_t)(nslots - 1);
  uint64_t win_m2 = 0, win_maxfast = 0;
  if (pipe) {
    nb = nslots >> region_bits;
    TORCH_CHECK(ev_packed2->numel() == ev_packed->numel(),
                "pipelined scatter buffers must match");
    magic_div_u64(len_ms, &win_m2, &win_maxfast);
    // Same scatter-variant/segmentation decision as radix_window_insert
    // (COUNT mode here).
    kind = scatter_kind_env();
    int coarse = scatter_coarse_bits(kind);
    seg_bits = (int)region_bits + coarse;
    nseg = nslots >> seg_bits;
    staged_lds =
        (size_t)nseg * SC_GRAN * 8 + 4 * (size_t)nseg * sizeof(int);
    while (kind == SCAT_STAGED &&
           (nseg < 1 || staged_lds > 160 * 1024 || seg_bits > 13)) {
      if (coarse > 0 && seg_bits > 13) {
        coarse -= 1;
      } else {
        kind = SCAT_FIXED;
        coarse = 0;
      }
      seg_bits = (int)region_bits + coarse;
      nseg = nslots >> seg_bits;
      staged_lds =
          (size_t)nseg * SC_GRAN * 8 + 4 * (size_t)nseg * sizeof(int);
    }
    if (seg_bits > 13) {
      seg_bits = (int)region_bits;
      nseg = nb;
    }
    cap = ev_packed->numel() / nseg;
    if (kind == SCAT_STAGED) {
      cap &= ~(int64_t)(SC_GRAN - 1);
      if (cap < SC_GRAN) {
        kind = SCAT_FIXED;
        seg_bits = (int)region_bits;
        nseg = nb;
        cap = ev_packed->numel() / nseg;
      }
    }
    agg_offsets = at::arange(
        nseg,
        at::TensorOptions().dtype(at::kInt).device(tkeys.device()));
    agg_offsets = agg_offsets * (int)cap;
    HIP_CHECK(hipStreamCreateWithFlags(&sc_stream, hipStreamNonBlocking));
    for (int i = 0; i < 2; ++i) {
      HIP_CHECK(hipEventCreateWithFlags(&ev_sc[i], hipEventDisableTiming));
      HIP_CHECK(hipEventCreateWithFlags(&ev_ag[i], hipEventDisableTiming));
    }
  }

  {
    pybind11::gil_scoped_release release;
    for (int64_t s = 0; s < n_steps; ++s) {
      if (step_ns != nullptr) {
        step_ns[s] = std::chrono::duration_cast<std::chrono::nanoseconds>(
                         std::chrono::steady_clock::now().time_since_epoch())
                         .count();
      }
      int64_t step = start_step + s;
      auto& keys = key_pool[step % pool];
      auto& ts = ts_pool[step % pool];
      int64_t base = align_ms + step * sim_ms_per_batch;
      if (use_radix_v2) {
        radix_v2_window_insert(
            keys, ts, cur_k, cur_v, max_ts, error_flag, *v2_gcur,
            *v2_gres, *v2_ev, *v2_ev_res, align_ms, len_ms, base,
            region_bits);
      } else if (pipe) {
        int par = (int)(step & 1);
        torch::Tensor& gcur = par ? *gcursors2 : *gcursors;
        torch::Tensor& evp = par ? *ev_packed2 : *ev_packed;
        torch::Tensor& ovc = par ? *ov_cursor2 : *ov_cursor;
        torch::Tensor& ovp = par ? *ov_packed2 : *ov_packed;
        if (s >= 2) {
          // Buffer reuse: the agg that read this parity's buffers two
          // steps ago must have drained them.
          HIP_CHECK(hipStreamWaitEvent(sc_stream, ev_ag[par], 0));
        }
        HIP_CHECK(hipMemsetAsync(
            gcur.data_ptr<int32_t>(), 0, (size_t)nseg * sizeof(int),
            sc_stream));
        HIP_CHECK(hipMemsetAsync(
            ovc.data_ptr<int32_t>(), 0, sizeof(int), sc_stream));
        auto scat_step = [&](auto tsptr) {
          using TSV =
              std::remove_const_t<std::remove_pointer_t<decltype(tsptr)>>;
          if (kind == SCAT_STAGED) {
            int su = 16;
            if (const char* e = std::getenv("BYTEWAX_SCATTER_U")) {
              int v = atoi(e);
              if (v == 8 || v == 16 || v == 32) su = v;
            }
            unsigned gs = (unsigned)((n + 256 * su - 1) / (256 * su));
            if (gs > 1024) gs = 1024;
            if (gs < 1) gs = 1;
            auto launch_staged = [&](auto kern) {
              hipLaunchKernelGGL(
                  kern, dim3(gs), block,
                  staged_lds, sc_stream, keys.data_ptr<int32_t>(),
                  tsptr, (const int64_t*)nullptr, n,
                  align_ms, len_ms, len_ms, base, mask, seg_bits, cap,
                  gcur.data_ptr<int32_t>(), (uint64_t*)evp.data_ptr<int64_t>(),
                  (int64_t*)nullptr, ovc.data_ptr<int32_t>(),
                  (uint64_t*)ovp.data_ptr<int64_t>(), (int64_t*)nullptr,
                  ovp.numel(),
                  (unsigned long long*)max_ts.data_ptr<int64_t>(),
                  error_flag.data_ptr<int32_t>(), win_m2, win_maxfast);
            };
            if (su == 8)
              launch_staged(k_radix_scatter_staged<AGG_COUNT, TSV, 8>);
            else if (su == 32)
              launch_staged(k_radix_scatter_staged<AGG_COUNT, TSV, 32>);
            else
              launch_staged(k_radix_scatter_staged<AGG_COUNT, TSV, 16>);
            // (pipe path keeps 256-thread blocks; serial is default)
          } else if (kind == SCAT_DIRECT) {
            hipLaunchKernelGGL(
                (k_radix_scatter_direct<AGG_COUNT, TSV>), grid, block, 0,
                sc_stream, keys.data_ptr<int32_t>(), tsptr,
                (const int64_t*)nullptr, n, align_ms, len_ms, len_ms, base,
                mask,
                seg_bits, cap, gcur.data_ptr<int32_t>(),
                (uint64_t*)evp.data_ptr<int64_t>(), (int64_t*)nullptr,
                ovc.data_ptr<int32_t>(), (uint64_t*)ovp.data_ptr<int64_t>(),
                (int64_t*)nullptr, ovp.numel(),
                (unsigned long long*)max_ts.data_ptr<int64_t>(),
                error_flag.data_ptr<int32_t>(), win_m2, win_maxfast);
          } else {
            size_t hist_lds = (size_t)nseg * sizeof(int);
            hipLaunchKernelGGL(
                (k_radix_scatter_fixed<AGG_COUNT, TSV>), grid, block,
                2 * hist_lds,
                sc_stream, keys.data_ptr<int32_t>(), tsptr,
                (const int64_t*)nullptr, n, align_ms, len_ms, len_ms, base, mask,
                seg_bits, cap, gcur.data_ptr<int32_t>(),
                (uint64_t*)evp.data_ptr<int64_t>(), (int64_t*)nullptr,
                ovc.data_ptr<int32_t>(), (uint64_t*)ovp.data_ptr<int64_t>(),
                (int64_t*)nullptr, ovp.numel(),
                (unsigned long long*)max_ts.data_ptr<int64_t>(),
                error_flag.data_ptr<int32_t>(), win_m2, win_maxfast);
          }
        };
        if (ts.scalar_type() == torch::kInt32)
          scat_step(ts.data_ptr<int32_t>());
        else scat_step(ts.data_ptr<int64_t>());
        HIP_CHECK(hipEventRecord(ev_sc[par], sc_stream));
        HIP_CHECK(hipStreamWaitEvent(stream, ev_sc[par], 0));
        size_t agg_lds = (size_t)16 << seg_bits;
        dim3 agg_block(seg_bits >= 12 ? 1024 : 256);
        hipLaunchKernelGGL(
            k_radix_agg<AGG_COUNT>, dim3((unsigned)nseg), agg_block,
            agg_lds, stream, (const uint64_t*)evp.data_ptr<int64_t>(),
            (const int64_t*)nullptr, agg_offsets.data_ptr<int32_t>(),
            gcur.data_ptr<int32_t>(), cap,
            (uint64_t*)cur_k.data_ptr<int64_t>(),
            (unsigned long long*)cur_v.data_ptr<int64_t>(), mask,
            (int)region_bits, seg_bits, error_flag.data_ptr<int32_t>());
        hipLaunchKernelGGL(
            k_overflow_agg<AGG_COUNT>, dim3(64), block, 0, stream,
            (const uint64_t*)ovp.data_ptr<int64_t>(),
            (const int64_t*)nullptr, ovc.data_ptr<int32_t>(), ovp.numel(),
            (uint64_t*)cur_k.data_ptr<int64_t>(),
            (unsigned long long*)cur_v.data_ptr<int64_t>(), mask,
            (int)region_bits, error_flag.data_ptr<int32_t>());
        HIP_CHECK(hipEventRecord(ev_ag[par], stream));
      } else if (use_radix) {
        radix_window_insert(
            keys, ts, c10::nullopt, cur_k, cur_v, max_ts, error_flag,
            *gcursors, *ev_packed, *ev_vals, *ov_cursor, *ov_packed,
            *ov_vals, align_ms, len_ms, AGG_COUNT, base, region_bits);
      } else {
        auto launch = [&](auto kern, auto tsptr) {
          hipLaunchKernelGGL(
              kern, grid, block, 0, stream, keys.data_ptr<int32_t>(),
              tsptr, (const int64_t*)nullptr, n,
              (uint64_t*)cur_k.data_ptr<int64_t>(),
              (unsigned long long*)cur_v.data_ptr<int64_t>(),
              (uint64_t)(nslots - 1), align_ms, len_ms, len_ms, base,
              (int)region_bits,
              (unsigned long long*)max_ts.data_ptr<int64_t>(),
              error_flag.data_ptr<int32_t>(), (const int64_t*)nullptr);
        };
        auto disp = [&](auto tsptr) {
          using TSV =
              std::remove_const_t<std::remove_pointer_t<decltype(tsptr)>>;
          if (dedup)
            launch(k_window_agg_insert<AGG_COUNT, true, TSV>, tsptr);
          else launch(k_window_agg_insert<AGG_COUNT, false, TSV>, tsptr);
        };
        if (ts.scalar_type() == torch::kInt32)
          disp(ts.data_ptr<int32_t>());
        else disp(ts.data_ptr<int64_t>());
      }

      int64_t wm = base + sim_ms_per_batch - 1;
      int64_t horizon = (wm - wait_ms - align_ms) / len_ms;
      if (horizon > closed_horizon) {
        HIP_CHECK(hipMemsetAsync(out_n.data_ptr<int32_t>(), 0, sizeof(int), stream));
        hipLaunchKernelGGL(
            k_close_migrate, dim3(n_blocks(nslots, 256)), block, 0, stream,
            (const uint64_t*)cur_k.data_ptr<int64_t>(),
            (const unsigned long long*)cur_v.data_ptr<int64_t>(), nslots,
            horizon, (uint64_t*)alt_k.data_ptr<int64_t>(),
            (unsigned long long*)alt_v.data_ptr<int64_t>(),
            (uint64_t)(nslots - 1), (int)region_bits,
            out_keys.data_ptr<int32_t>(), out_wins.data_ptr<int32_t>(),
            out_vals.data_ptr<int64_t>(), out_n.data_ptr<int32_t>(),
            out_keys.numel(), error_flag.data_ptr<int32_t>());
        HIP_CHECK(hipMemcpyAsync(h_n, out_n.data_ptr<int32_t>(), sizeof(int),
                       hipMemcpyDeviceToHost, stream));
        std::swap(cur_k, alt_k);
        std::swap(cur_v, alt_v);
        swap_parity ^= 1;
        // Reset the retired table asynchronously for the next close.
        HIP_CHECK(hipMemsetAsync(alt_k.data_ptr<int64_t>(), 0xFF,
                                 (size_t)nslots * 8, stream));
        HIP_CHECK(hipMemsetAsync(alt_v.data_ptr<int64_t>(), 0,
                                 (size_t)nslots * 8, stream));
        HIP_CHECK(hipStreamSynchronize(stream));
        total_rows += *h_n;
        closed_horizon = horizon;
      }
    }
    if (sc_stream != nullptr) {
      HIP_CHECK(hipStreamSynchronize(sc_stream));
    }
    HIP_CHECK(hipStreamSynchronize(stream));
  }
  if (sc_stream != nullptr) {
    for (int i = 0; i < 2; ++i) {
      HIP_CHECK(hipEventDestroy(ev_sc[i]));
      HIP_CHECK(hipEventDestroy(ev_ag[i]));
    }
    HIP_CHECK(hipStreamDestroy(sc_stream));
  }
  HIP_CHECK(hipHostFree(h_n));
  auto* st = state_out.data_ptr<int64_t>();
  st[0] = closed_horizon;
  st[1] = total_rows;
  st[2] = swap_parity;
  return total_rows;
}

// hipGraph variant of the native step loop: one pool cycle of
// {insert, ts-bump} kernels is captured once and replayed per group —
// the host cost of a whole pool of steps collapses to one
// hipGraphLaunch.  Timestamps advance via the device-side offset
// scalar so the captured graph stays valid across replays.  Window
// closes happen between replays (their steps are arithmetic).
// COUNT mode, single-pass insert only (the latency-oriented path).
int64_t native_run_window_steps_graph(
    std::vector<torch::Tensor> key_pool,
    std::vector<torch::Tensor> ts_pool,
    int64_t start_step,
    int64_t n_steps,
    int64_t sim_ms_per_batch,
    torch::Tensor tkeys,
    torch::Tensor tvals,
    torch::Tensor max_ts,
    torch::Tensor error_flag,
    torch::Tensor out_keys,
    torch::Tensor out_wins,
    torch::Tensor out_vals,
    torch::Tensor out_n,
    int64_t align_ms,
    int64_t len_ms,
    int64_t wait_ms,
    int64_t closed_horizon_in,
    torch::Tensor state_out,  // int64 CPU [3]
    int64_t region_bits,
    torch::Tensor alt_tkeys,
    torch::Tensor alt_tvals,
    torch::Tensor ts_base_dev) {  // int64 device scalar
  TORCH_CHECK(!key_pool.empty(), "empty key pool");
  int64_t nslots = tkeys.numel();
  TORCH_CHECK((nslots & (nslots - 1)) == 0, "table size must be 2^k");
  int64_t n = key_pool[0].numel();
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(n, 256));
  int pool = (int)key_pool.size();
  uint64_t mask = (uint64_t)(nslots - 1);

  // The device offset starts at the first step's base.
  ts_base_dev.fill_(align_ms + start_step * sim_ms_per_batch);

  auto launch_insert_on = [&](int j, hipStream_t st) {
    hipLaunchKernelGGL(
        (k_window_agg_insert<AGG_COUNT, false>), grid, block, 0, st,
        key_pool[j].data_ptr<int32_t>(), ts_pool[j].data_ptr<int64_t>(),
        (const int64_t*)nullptr, n, (uint64_t*)tkeys.data_ptr<int64_t>(),
        (unsigned long long*)tvals.data_ptr<int64_t>(), mask, align_ms,
        len_ms, len_ms, 0, (int)region_bits,
        (unsigned long long*)max_ts.data_ptr<int64_t>(),
        error_flag.data_ptr<int32_t>(),
        (const int64_t*)ts_base_dev.data_ptr<int64_t>());
    hipLaunchKernelGGL(
        k_bump, dim3(1), dim3(1), 0, st,
        ts_base_dev.data_ptr<int64_t>(), sim_ms_per_batch);
  };
  auto launch_insert = [&](int j) { launch_insert_on(j, stream); };

  // NOTE: inserts always target `tkeys`/`tvals` (the tensors captured
  // into the graph), so closes must migrate live cells back into the
  // SAME tensors — a double swap: cur -> alt -> cur with a reset in
  // between keeps probe-chain hygiene while preserving buffer
  // identity for the graph.
  int64_t closed_horizon = closed_horizon_in;
  int64_t total_rows = 0;
  int* h_n = nullptr;
  HIP_CHECK(hipHostMalloc((void**)&h_n, sizeof(int), hipHostMallocDefault));

  hipGraph_t graph = nullptr;
  hipGraphExec_t gexec = nullptr;
  {
    pybind11::gil_scoped_release release;
    // Capture must happen on a non-default stream; the replays run on
    // the torch stream.
    hipStream_t cs = nullptr;
    HIP_CHECK(hipStreamCreateWithFlags(&cs, hipStreamNonBlocking));
    HIP_CHECK(hipStreamBeginCapture(cs, hipStreamCaptureModeThreadLocal));
    for (int j = 0; j < pool; ++j) launch_insert_on(j, cs);
    HIP_CHECK(hipStreamEndCapture(cs, &graph));
    HIP_CHECK(hipStreamDestroy(cs));
    HIP_CHECK(hipGraphInstantiate(&gexec, graph, nullptr, nullptr, 0));

    auto horizon_after = [&](int64_t step) {
      int64_t wm = align_ms + (step + 1) * sim_ms_per_batch - 1;
      return (wm - wait_ms - align_ms) / len_ms;
    };
    auto do_close = [&](int64_t horizon) {
      HIP_CHECK(hipMemsetAsync(out_n.data_ptr<int32_t>(), 0, sizeof(int),
                               stream));
      // Migrate live cells out and back so the graph's captured table
      // pointers stay authoritative.
      hipLaunchKernelGGL(
          k_close_migrate, dim3(n_blocks(nslots, 256)), block, 0, stream,
          (const uint64_t*)tkeys.data_ptr<int64_t>(),
          (const unsigned long long*)tvals.data_ptr<int64_t>(), nslots,
          horizon, (uint64_t*)alt_tkeys.data_ptr<int64_t>(),
          (unsigned long long*)alt_tvals.data_ptr<int64_t>(), mask,
          (int)region_bits, out_keys.data_ptr<int32_t>(),
          out_wins.data_ptr<int32_t>(), out_vals.data_ptr<int64_t>(),
          out_n.data_ptr<int32_t>(), out_keys.numel(),
          error_flag.data_ptr<int32_t>());
      HIP_CHECK(hipMemcpyAsync(h_n, out_n.data_ptr<int32_t>(), sizeof(int),
                               hipMemcpyDeviceToHost, stream));
      // Reset the primary and migrate back (no emissions: horizon
      // below everything live now).
      HIP_CHECK(hipMemsetAsync(tkeys.data_ptr<int64_t>(), 0xFF,
                               (size_t)nslots * 8, stream));
      HIP_CHECK(hipMemsetAsync(tvals.data_ptr<int64_t>(), 0,
                               (size_t)nslots * 8, stream));
      hipLaunchKernelGGL(
          k_close_migrate, dim3(n_blocks(nslots, 256)), block, 0, stream,
          (const uint64_t*)alt_tkeys.data_ptr<int64_t>(),
          (const unsigned long long*)alt_tvals.data_ptr<int64_t>(), nslots,
          (int64_t)(-(1LL << 40)), (uint64_t*)tkeys.data_ptr<int64_t>(),
          (unsigned long long*)tvals.data_ptr<int64_t>(), mask,
          (int)region_bits, out_keys.data_ptr<int32_t>(),
          out_wins.data_ptr<int32_t>(), out_vals.data_ptr<int64_t>(),
          out_n.data_ptr<int32_t>(), out_keys.numel(),
          error_flag.data_ptr<int32_t>());
      HIP_CHECK(hipMemsetAsync(alt_tkeys.data_ptr<int64_t>(), 0xFF,
                               (size_t)nslots * 8, stream));
      HIP_CHECK(hipMemsetAsync(alt_tvals.data_ptr<int64_t>(), 0,
                               (size_t)nslots * 8, stream));
      HIP_CHECK(hipStreamSynchronize(stream));
      total_rows += *h_n;
      closed_horizon = horizon;
    };

    int64_t s = 0;
    while (s < n_steps) {
      bool group_ok = (n_steps - s) >= pool &&
                      horizon_after(start_step + s + pool - 1) ==
                          closed_horizon;
      if (group_ok) {
        HIP_CHECK(hipGraphLaunch(gexec, stream));
        s += pool;
      } else {
        launch_insert((int)((start_step + s) % pool));
        int64_t h = horizon_after(start_step + s);
        s += 1;
        if (h > closed_horizon) do_close(h);
      }
    }
    HIP_CHECK(hipStreamSynchronize(stream));
  }
  HIP_CHECK(hipGraphExecDestroy(gexec));
  HIP_CHECK(hipGraphDestroy(graph));
  HIP_CHECK(hipHostFree(h_n));
  auto* st = state_out.data_ptr<int64_t>();
  st[0] = closed_horizon;
  st[1] = total_rows;
  st[2] = 0;  // table identity preserved
  return total_rows;
}

// Fused radix session insert: AGG_TS scatter -> per-segment LDS
// session aggregation -> sequential overflow walk.  One call per
// batch; emissions accumulate in the out buffers until drained.
void session_radix_insert(
    torch::Tensor keys,
    torch::Tensor ts,  // int32 template or int64 absolute
    int64_t ts_base,
    int64_t gap_ms,
    torch::Tensor skeys,
    torch::Tensor sstart,
    torch::Tensor slast,
    torch::Tensor sacc,
    torch::Tensor gcursors,
    torch::Tensor ev_packed,
    torch::Tensor ev_vals,
    torch::Tensor ov_cursor,
    torch::Tensor ov_packed,
    torch::Tensor ov_vals,
    torch::Tensor out_keys,
    torch::Tensor out_start,
    torch::Tensor out_end,
    torch::Tensor out_vals,
    torch::Tensor out_n,
    torch::Tensor max_ts,
    torch::Tensor error_flag,
    int64_t seg_bits) {
  check_dev(keys, torch::kInt32, "keys");
  bool ts32 = ts.scalar_type() == torch::kInt32;
  if (!ts32) check_dev(ts, torch::kInt64, "ts");
  int64_t n = keys.numel();
  int64_t nslots = skeys.numel();
  TORCH_CHECK((nslots & (nslots - 1)) == 0, "table size must be 2^k");
  TORCH_CHECK(seg_bits > 0 && seg_bits < 40, "bad seg_bits");
  int64_t nseg = nslots >> seg_bits;
  TORCH_CHECK(nseg >= 1 && nseg <= 8192, "segment count out of range");
  TORCH_CHECK(gcursors.numel() >= nseg, "gcursors too small");
  int64_t cap = ev_packed.numel() / nseg;
  bool staged = true;
  cap &= ~(int64_t)(SC_GRAN - 1);
  if (cap < SC_GRAN) {
    staged = false;
    cap = ev_packed.numel() / nseg;
  }
  TORCH_CHECK(cap * nseg >= 2 * n || cap >= n,
              "session scatter buffers too small");
  TORCH_CHECK(ev_vals.numel() >= nseg * cap, "ev_vals too small");
  if (n == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  uint64_t mask = (uint64_t)(nslots - 1);
  gcursors.narrow(0, 0, nseg).zero_();
  ov_cursor.zero_();
  uint64_t win_m2, win_maxfast;
  magic_div_u64((int64_t)1 << 40, &win_m2, &win_maxfast);
  size_t staged_lds = (size_t)nseg * SC_GRAN * 8 * 2 +
                      4 * (size_t)nseg * sizeof(int);
  auto scat512 = [&](auto kern, auto tsptr, unsigned gx, size_t lds) {
    hipLaunchKernelGGL(
        kern, dim3(gx), dim3(512), lds, stream,
        keys.data_ptr<int32_t>(), tsptr, (const int64_t*)nullptr, n, 0,
        (int64_t)1 << 40, (int64_t)1 << 40, ts_base, mask, (int)seg_bits,
        cap,
        gcursors.data_ptr<int32_t>(),
        (uint64_t*)ev_packed.data_ptr<int64_t>(),
        ev_vals.data_ptr<int64_t>(), ov_cursor.data_ptr<int32_t>(),
        (uint64_t*)ov_packed.data_ptr<int64_t>(),
        ov_vals.data_ptr<int64_t>(), ov_packed.numel(),
        (unsigned long long*)max_ts.data_ptr<int64_t>(),
        error_flag.data_ptr<int32_t>(), win_m2, win_maxfast);
  };
  auto scat = [&](auto kern, auto tsptr, unsigned gx, size_t lds) {
    hipLaunchKernelGGL(
        kern, dim3(gx), dim3(256), lds, stream,
        keys.data_ptr<int32_t>(), tsptr, (const int64_t*)nullptr, n, 0,
        (int64_t)1 << 40, (int64_t)1 << 40, ts_base, mask, (int)seg_bits, cap,
        gcursors.data_ptr<int32_t>(),
        (uint64_t*)ev_packed.data_ptr<int64_t>(),
        ev_vals.data_ptr<int64_t>(), ov_cursor.data_ptr<int32_t>(),
        (uint64_t*)ov_packed.data_ptr<int64_t>(),
        ov_vals.data_ptr<int64_t>(), ov_packed.numel(),
        (unsigned long long*)max_ts.data_ptr<int64_t>(),
        error_flag.data_ptr<int32_t>(), win_m2, win_maxfast);
  };
  auto scat_any = [&](auto tsptr) {
    using TSV = std::remove_const_t<std::remove_pointer_t<decltype(tsptr)>>;
    if (staged && staged_lds <= 160 * 1024) {
      unsigned gs = (unsigned)((n + 8191) / 8192);
      if (gs > 1024) gs = 1024;
      if (gs < 1) gs = 1;
      scat512(k_radix_scatter_staged<AGG_TS, TSV, 16, 512>, tsptr, gs,
              staged_lds);
    } else {
      size_t hist_lds = (size_t)nseg * sizeof(int);
      scat(k_radix_scatter_fixed<AGG_TS, TSV>, tsptr,
           (unsigned)n_blocks(n, 256), 2 * hist_lds);
    }
  };
  if (ts32) scat_any(ts.data_ptr<int32_t>());
  else scat_any(ts.data_ptr<int64_t>());

  auto offsets = at::arange(
      nseg, at::TensorOptions().dtype(at::kInt).device(keys.device()));
  offsets = offsets * (int)cap;
  // LDS staging sized ~2x the expected distinct keys per segment.
  int lds_bits = 12;
  while ((int64_t)1 << lds_bits > nslots >> 1 && lds_bits > 6) lds_bits--;
  size_t agg_lds = (size_t)32 << lds_bits;
  hipLaunchKernelGGL(
      k_radix_session_agg, dim3((unsigned)nseg),
      dim3(lds_bits >= 12 ? 1024 : 256), agg_lds, stream,
      (const uint64_t*)ev_packed.data_ptr<int64_t>(),
      ev_vals.data_ptr<int64_t>(), offsets.data_ptr<int32_t>(),
      gcursors.data_ptr<int32_t>(), cap, lds_bits,
      (uint64_t*)skeys.data_ptr<int64_t>(),
      (long long*)sstart.data_ptr<int64_t>(),
      (long long*)slast.data_ptr<int64_t>(),
      (long long*)sacc.data_ptr<int64_t>(), mask, gap_ms,
      ov_cursor.data_ptr<int32_t>(),
      (uint64_t*)ov_packed.data_ptr<int64_t>(),
      ov_vals.data_ptr<int64_t>(), ov_packed.numel(),
      out_keys.data_ptr<int32_t>(), out_start.data_ptr<int64_t>(),
      out_end.data_ptr<int64_t>(), out_vals.data_ptr<int64_t>(),
      out_n.data_ptr<int32_t>(), out_keys.numel(),
      error_flag.data_ptr<int32_t>());
  hipLaunchKernelGGL(
      k_session_ov_walk, dim3(1), dim3(64), 0, stream,
      (const uint64_t*)ov_packed.data_ptr<int64_t>(),
      ov_vals.data_ptr<int64_t>(), ov_cursor.data_ptr<int32_t>(),
      ov_packed.numel(), (uint64_t*)skeys.data_ptr<int64_t>(),
      (long long*)sstart.data_ptr<int64_t>(),
      (long long*)slast.data_ptr<int64_t>(),
      (long long*)sacc.data_ptr<int64_t>(), mask, gap_ms,
      out_keys.data_ptr<int32_t>(), out_start.data_ptr<int64_t>(),
      out_end.data_ptr<int64_t>(), out_vals.data_ptr<int64_t>(),
      out_n.data_ptr<int32_t>(), out_keys.numel(),
      error_flag.data_ptr<int32_t>());
}

// ---- String dictionary wrappers ----

int64_t dict_encode(
    torch::Tensor bytes,    // uint8 device [total_bytes]
    torch::Tensor offs,     // int64 device [n+1]
    torch::Tensor dlo,      // int64 device [nslots] (init EMPTY)
    torch::Tensor dhi,      // int64 device [nslots]
    torch::Tensor dids,     // int32 device [nslots]
    torch::Tensor counter,  // int32 device [1]
    torch::Tensor new_ids,  // int32 device [new_cap]
    torch::Tensor new_idx,  // int32 device [new_cap]
    torch::Tensor new_n,    // int32 device [1]
    torch::Tensor out_ids,  // int32 device [n]
    torch::Tensor error_flag) {
  check_dev(bytes, torch::kUInt8, "bytes");
  check_dev(offs, torch::kInt64, "offs");
  check_dev(dlo, torch::kInt64, "dlo");
  check_dev(out_ids, torch::kInt32, "out_ids");
  int64_t n = offs.numel() - 1;
  int64_t nslots = dlo.numel();
  TORCH_CHECK((nslots & (nslots - 1)) == 0, "dict size must be 2^k");
  if (n == 0) return 0;
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(n, 256));
  uint64_t mask = (uint64_t)(nslots - 1);
  new_n.zero_();
  hipLaunchKernelGGL(
      k_dict_insert, grid, block, 0, stream,
      bytes.data_ptr<uint8_t>(), offs.data_ptr<int64_t>(), n,
      (uint64_t*)dlo.data_ptr<int64_t>(),
      (uint64_t*)dhi.data_ptr<int64_t>(), dids.data_ptr<int32_t>(), mask,
      counter.data_ptr<int32_t>(), new_ids.data_ptr<int32_t>(),
      new_idx.data_ptr<int32_t>(), new_n.data_ptr<int32_t>(),
      new_ids.numel(), error_flag.data_ptr<int32_t>());
  hipLaunchKernelGGL(
      k_dict_lookup, grid, block, 0, stream,
      bytes.data_ptr<uint8_t>(), offs.data_ptr<int64_t>(), n,
      (const uint64_t*)dlo.data_ptr<int64_t>(),
      (const uint64_t*)dhi.data_ptr<int64_t>(), dids.data_ptr<int32_t>(),
      mask, out_ids.data_ptr<int32_t>(), error_flag.data_ptr<int32_t>());
  return n;
}

void dict_restore(
    torch::Tensor bytes,
    torch::Tensor offs,
    torch::Tensor ids,  // int32 device [n] pinned id per string
    torch::Tensor dlo,
    torch::Tensor dhi,
    torch::Tensor dids,
    torch::Tensor error_flag) {
  check_dev(bytes, torch::kUInt8, "bytes");
  check_dev(offs, torch::kInt64, "offs");
  check_dev(ids, torch::kInt32, "ids");
  int64_t n = offs.numel() - 1;
  int64_t nslots = dlo.numel();
  TORCH_CHECK((nslots & (nslots - 1)) == 0, "dict size must be 2^k");
  if (n == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(
      k_dict_insert_pinned, dim3(n_blocks(n, 256)), dim3(256), 0, stream,
      bytes.data_ptr<uint8_t>(), offs.data_ptr<int64_t>(),
      ids.data_ptr<int32_t>(), n, (uint64_t*)dlo.data_ptr<int64_t>(),
      (uint64_t*)dhi.data_ptr<int64_t>(), dids.data_ptr<int32_t>(),
      (uint64_t)(nslots - 1), error_flag.data_ptr<int32_t>());
}

// Pack a str batch into per-destination-rank wire segments for the
// all-to-allv string exchange (owner = content hash % world).
// Returns nothing; fills `counts` (strings/rank), `byte_counts`
// (bytes/rank), `send_lens`/`send_ts`/`send_vals` (bucket-major wire
// order) and `send_bytes`.  All device work stays on the current
// stream (cumsums included); callers read counts to host for the
// collective split sizes.
void str_exchange_pack(
    torch::Tensor bytes,   // uint8 [total]
    torch::Tensor offs,    // int64 [n+1]
    torch::Tensor ts,      // int64 [n]
    c10::optional<torch::Tensor> vals,  // int64 [n]
    int64_t world,
    torch::Tensor counts,       // int32 [world] out (zeroed here)
    torch::Tensor byte_counts,  // int64 [world] out (zeroed here)
    torch::Tensor send_lens,    // int32 [n] out
    torch::Tensor send_ts,      // int64 [n] out
    torch::Tensor send_vals,    // int64 [n or 0] out
    torch::Tensor send_bytes    // uint8 [total] out
) {
  check_dev(bytes, torch::kUInt8, "bytes");
  check_dev(offs, torch::kInt64, "offs");
  check_dev(ts, torch::kInt64, "ts");
  check_dev(send_lens, torch::kInt32, "send_lens");
  int64_t n = offs.numel() - 1;
  TORCH_CHECK(world >= 1, "world must be >= 1");
  TORCH_CHECK(send_lens.numel() >= n, "send_lens too small");
  TORCH_CHECK(send_bytes.numel() >= bytes.numel(), "send_bytes too small");
  counts.zero_();
  byte_counts.zero_();
  if (n == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  dim3 block(256);
  dim3 grid(n_blocks(n, 256));
  hipLaunchKernelGGL(
      k_str_bucket_hist, grid, block, 0, stream,
      bytes.data_ptr<uint8_t>(), offs.data_ptr<int64_t>(), n, (int)world,
      counts.data_ptr<int32_t>(),
      (long long*)byte_counts.data_ptr<int64_t>());
  auto cursors =
      (torch::cumsum(counts, 0, torch::kInt32) - counts).contiguous();
  const int64_t* vptr =
      vals.has_value() ? vals->data_ptr<int64_t>() : nullptr;
  auto src_idx = torch::empty(
      {n}, torch::TensorOptions()
               .dtype(torch::kInt32)
               .device(bytes.device()));
  hipLaunchKernelGGL(
      k_str_meta_scatter, grid, block, 0, stream,
      bytes.data_ptr<uint8_t>(), offs.data_ptr<int64_t>(),
      ts.data_ptr<int64_t>(), vptr, n, (int)world,
      cursors.data_ptr<int32_t>(), send_lens.data_ptr<int32_t>(),
      send_ts.data_ptr<int64_t>(),
      vptr != nullptr ? send_vals.data_ptr<int64_t>() : nullptr,
      src_idx.data_ptr<int32_t>());
  auto lens64 = send_lens.narrow(0, 0, n).to(torch::kInt64);
  auto send_offs = (torch::cumsum(lens64, 0) - lens64).contiguous();
  hipLaunchKernelGGL(
      k_str_byte_gather, grid, block, 0, stream,
      bytes.data_ptr<uint8_t>(), offs.data_ptr<int64_t>(),
      src_idx.data_ptr<int32_t>(), send_offs.data_ptr<int64_t>(), n,
      send_bytes.data_ptr<uint8_t>());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("str_exchange_pack", &str_exchange_pack,
        "pack a str batch into per-rank wire segments (content-hash "
        "owner) for the all-to-allv string exchange");
  m.def("session_radix_insert", &session_radix_insert,
        "fused radix session insert: AGG_TS scatter + per-segment LDS "
        "session aggregation + sequential overflow walk");
  m.def("dict_encode", &dict_encode,
        "device string-dictionary encode: 2-phase insert+lookup");
  m.def("dict_restore", &dict_restore,
        "device string-dictionary restore with pinned ids");
  m.def("window_agg_insert", &window_agg_insert,
        "Fused window-id + hash-insert + watermark over an event batch");
  m.def("radix_v2_window_insert", &radix_v2_window_insert,
        "Experimental two-level radix with full-line LDS-staged "
        "scatter (COUNT mode)");
  m.def("radix_scatter_only", &radix_scatter_only,
        "Scatter stage of the radix insert on the current stream");
  m.def("radix_agg_only", &radix_agg_only,
        "Aggregation stage of the radix insert on the current stream");
  m.def("radix_window_insert", &radix_window_insert,
        "Radix-partitioned LDS-staged keyed window aggregation");
  m.def("close_migrate", &close_migrate,
        "Window close with reclamation: emit closed cells and migrate "
        "live cells into a fresh table");
  m.def("close_extract", &close_extract,
        "Extract (and clear) closed windows from the keyed state table");
  m.def("stats_insert", &stats_insert,
        "Keyed running count/sum/min/max over an event batch (1BRC)");
  m.def("radix_stats_insert", &radix_stats_insert,
        "Radix-partitioned LDS-staged keyed stats aggregation");
  m.def("stats_fixup", &stats_fixup,
        "Add count/sum deltas to existing stats slots (recovery)");
  m.def("stats_extract", &stats_extract,
        "Extract (and clear) keyed stats below a window horizon");
  m.def("radix_join_insert", &radix_join_insert,
        "Region-partitioned stream join insert (L2-local table ops)");
  m.def("join_insert", &join_insert,
        "Stream-stream hash join insert; emits completed pairs");
  m.def("session_insert", &session_insert,
        "Gap-based session aggregation over a (key, ts)-sorted batch");
  m.def("session_close_migrate", &session_close_migrate,
        "Emit sessions idle past the horizon; migrate live cells");
  m.def("session_restore", &session_restore,
        "Rebuild session cells from a host spill");
  m.def("session_merge_batch", &session_merge_batch,
        "Merge per-key per-batch (count, min_ts, max_ts) rows into "
        "the session table (gap >= batch span fast path)");
  m.def("join_extract", &join_extract,
        "Extract live join state (recovery snapshot)");
  m.def("filter_compact", &filter_compact,
        "Stream compaction of a RecordBatch by a boolean mask");
  m.def("bucket_hist", &bucket_hist, "Per-destination counts for exchange");
  m.def("bucket_scatter", &bucket_scatter,
        "Scatter events into per-destination segments for all-to-allv");
  m.def("native_run_window_steps_graph", &native_run_window_steps_graph,
        "hipGraph-captured native step loop (latency mode: one graph "
        "launch per pool cycle)");
  m.def("native_run_window_steps", &native_run_window_steps,
        "Run N steps of the columnar window pipeline with no Python "
        "between steps (native step loop)");
}
