// persia_amd HIP kernels (gfx950 / CDNA4, wave64).
//
// MI355X-native replacements for the reference's CPU hot loops:
//  * set-associative HBM hash table  <- EvictionMap LRU + per-sign locks
//    (rust/persia-embedding-holder/src/eviction_map.rs,
//     embedding_parameter_service/mod.rs:162-262)
//  * fused gather + segment-sum      <- add_assign_avx2 summation
//    (embedding_worker_service/mod.rs:486-629, persia-simd lib.rs:4-18)
//  * ordered gradient scatter        <- per-sign HashMap grad accumulation
//    (embedding_worker_service/mod.rs:782-814)
//  * fused row-wise sparse optimizers + weight bound
//    (persia-simd lib.rs:21-251, exact-math variants)
//
// Layout: keys u64[n_slots] (0 = empty), ticks u32[n_slots], arena
// f32[n_slots, row_width], row = [emb(dim) | opt_state].  Buckets of 8
// slots (one 64B cache line of keys); probe scans 4 consecutive buckets.
//
// Concurrency contract (matches the engine's stream discipline):
//  * keys within one lookup/update call are unique (deduped upstream);
//  * lookup calls are serialized on one stream (ForwardPipeline);
//  * an update may overlap a lookup; claims use atomicCAS on the key word,
//    so the only hazard is an eviction racing an in-flight update of the
//    evicted row — bounded-staleness noise the PERSIA algorithm tolerates
//    (SURVEY §7 hard part 2); it cannot corrupt the table structure.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hipcub/hipcub.hpp>

#include "common.h"

using ull = unsigned long long;

namespace {

__device__ __forceinline__ float pa_to_float(float x) { return x; }
__device__ __forceinline__ float pa_to_float(__half x) { return __half2float(x); }
__device__ __forceinline__ float pa_to_float(__hip_bfloat16 x) {
  return __bfloat162float(x);
}

// ---------------------------------------------------------------- store probe

// Find the slot holding key k, or -1.  Bucket loads hit one cache line.
__device__ __forceinline__ long long probe_find(const ull* __restrict__ keys,
                                                int64_t n_buckets, ull k) {
  const int64_t mask = n_buckets - 1;
  int64_t b = (int64_t)(k & (ull)mask);
  for (int p = 0; p < PA_PROBE_BUCKETS; ++p) {
    const int64_t base = ((b + p) & mask) * PA_BUCKET_SIZE;
#pragma unroll
    for (int s = 0; s < PA_BUCKET_SIZE; ++s) {
      if (keys[base + s] == k) return base + s;
    }
  }
  return -1;
}

// Probe-or-claim with bounded-window LRU eviction (store.py _probe_or_claim).
// Returns slot; *is_new = 1 when this thread claimed it (must init the row).
// Rows touched at the CURRENT tick are never eviction victims, so concurrent
// claims within one batch cannot evict each other; when the whole window is
// current-tick the key overflows to a miss (bounded churn).
__device__ long long probe_claim(ull* __restrict__ keys,
                                 unsigned* __restrict__ ticks,
                                 int64_t n_buckets, ull k, unsigned cur_tick,
                                 int* is_new, ull* victim_out) {
  const int64_t mask = n_buckets - 1;
  const int64_t b = (int64_t)(k & (ull)mask);
  for (int attempt = 0; attempt < 16; ++attempt) {
    long long empty = -1;
    long long victim = -1;
    unsigned victim_tick = 0xFFFFFFFFu;
    ull victim_key = 0;
    for (int p = 0; p < PA_PROBE_BUCKETS; ++p) {
      const int64_t base = ((b + p) & mask) * PA_BUCKET_SIZE;
#pragma unroll
      for (int s = 0; s < PA_BUCKET_SIZE; ++s) {
        const int64_t i = base + s;
        const ull ki = keys[i];
        if (ki == k) { *is_new = 0; return i; }
        if (ki == PA_EMPTY_KEY) {
          if (empty < 0) empty = i;
        } else {
          const unsigned t = ticks[i];
          if (t != cur_tick && (victim < 0 || t < victim_tick)) {
            victim = i; victim_tick = t; victim_key = ki;
          }
        }
      }
    }
    if (empty >= 0) {
      if (atomicCAS((ull*)&keys[empty], PA_EMPTY_KEY, k) == PA_EMPTY_KEY) {
        ticks[empty] = cur_tick;  // publish before peers pick victims
        *is_new = 1; return empty;
      }
      continue;  // lost the race; rescan
    }
    if (victim >= 0) {
      if (atomicCAS((ull*)&keys[victim], victim_key, k) == victim_key) {
        ticks[victim] = cur_tick;
        *is_new = 1;
        if (victim_out) *victim_out = victim_key;
        return victim;
      }
      continue;
    }
    break;  // every slot is current-tick: overflow
  }
  *is_new = 0;
  return -1;  // give up: treated as a non-admitted miss
}

// Eviction log (host-DRAM spill tier): a claim that evicts appends the
// victim's key to evict_keys and remembers the log index so init_gather can
// copy the victim ROW out before overwriting it — the engine drains the log
// into the host tier (SURVEY §7.7 spill design).
__global__ void probe_claim_kernel(ull* __restrict__ table_keys,
                                   unsigned* __restrict__ ticks,
                                   const ull* __restrict__ query,
                                   int64_t n, int64_t n_buckets, int train,
                                   unsigned tick, float admit_prob,
                                   long long* __restrict__ out_slot,
                                   int* __restrict__ out_new,
                                   ull* __restrict__ evict_keys,
                                   int* __restrict__ evict_count,
                                   long long* __restrict__ out_evict_idx,
                                   const long long* __restrict__ n_dev) {
  // n_dev: device-side element count (padded-dedup path — the true unique
  // count is produced on-GPU so the host never synchronizes); n is the
  // grid-sizing upper bound
  if (n_dev) n = *n_dev;
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const ull k = query[i];
  if (k == PA_EMPTY_KEY) {
    // a2a padding sentinel (distributed recv buffers): never probed — the
    // empty key would match every free slot
    out_slot[i] = -1;
    out_new[i] = 0;
    if (out_evict_idx) out_evict_idx[i] = -1;
    return;
  }
  long long slot;
  int is_new = 0;
  ull victim = 0;
  if (train) {
    bool admitted = true;
    if (admit_prob < 1.0f) {
      // stochastic admission re-rolled per tick (store.py _admitted)
      const double u = pa_u01(pa_splitmix64(k ^ (ull)tick));
      admitted = u < (double)admit_prob;
    }
    if (admitted) {
      slot = probe_claim(table_keys, ticks, n_buckets, k, tick, &is_new,
                         evict_keys ? &victim : nullptr);
    } else {
      slot = probe_find(table_keys, n_buckets, k);
    }
  } else {
    slot = probe_find(table_keys, n_buckets, k);
  }
  if (slot >= 0) ticks[slot] = tick;
  out_slot[i] = slot;
  out_new[i] = is_new;
  long long eidx = -1;
  if (evict_keys && victim != 0) {
    eidx = (long long)atomicAdd(evict_count, 1);
    evict_keys[eidx] = victim;
  }
  if (out_evict_idx) out_evict_idx[i] = eidx;
}

// one wave per key: init freshly claimed rows, then gather emb -> out.
// OutT = f32 (local path) or f16 (distributed wire: the gathered rows go
// straight onto the xGMI all-to-all, so the wire cast is fused here).
__device__ __forceinline__ void pa_store_out(float v, float* p) { *p = v; }
__device__ __forceinline__ void pa_store_out(float v, __half* p) {
  *p = __float2half(v);
}
template <typename OutT>
__global__ void init_gather_kernel(float* __restrict__ arena,
                                   const ull* __restrict__ query,
                                   const long long* __restrict__ slots,
                                   const int* __restrict__ is_new,
                                   OutT* __restrict__ out, int64_t n, int dim,
                                   int row_width, double lo, double hi,
                                   float state_init,
                                   const long long* __restrict__ evict_idx,
                                   float* __restrict__ evict_rows,
                                   const long long* __restrict__ n_dev) {
  if (n_dev) n = *n_dev;
  const int wave = threadIdx.x / PA_WAVE;
  const int lane = threadIdx.x % PA_WAVE;
  const int waves_per_block = blockDim.x / PA_WAVE;
  const int G = (dim < PA_WAVE && (PA_WAVE % dim) == 0) ? PA_WAVE / dim : 1;
  const int sub = (G > 1) ? lane / dim : 0;
  const int c0 = (G > 1) ? lane % dim : lane;
  const int64_t n_waveitems = (n + G - 1) / G;
  for (int64_t w = (int64_t)blockIdx.x * waves_per_block + wave;
       w < n_waveitems; w += (int64_t)gridDim.x * waves_per_block) {
    const int64_t i = w * G + sub;
    if (i >= n) continue;
    const long long slot = slots[i];
    OutT* dst = out + i * dim;
    if (slot < 0) {
      for (int c = c0; c < dim; c += PA_WAVE) pa_store_out(0.0f, dst + c);
      continue;
    }
    float* row = arena + (int64_t)slot * row_width;
    const int st = (G > 1) ? dim : PA_WAVE;  // subgroup stride
    if (evict_rows && is_new[i] && evict_idx[i] >= 0) {
      // spill: save the victim's full row before overwriting it
      float* spill = evict_rows + evict_idx[i] * row_width;
      for (int c = c0; c < row_width; c += st) spill[c] = row[c];
    }
    if (is_new[i]) {
      const uint64_t sign = pa_splitmix64_inv((uint64_t)query[i]);
      const uint64_t seed = pa_init_seed(sign);
      for (int c = c0; c < dim; c += st)
        row[c] = pa_init_val(seed, c, lo, hi);
      for (int c = dim + c0; c < row_width; c += st)
        row[c] = state_init;
    }
    for (int c = c0; c < dim; c += st) pa_store_out(row[c], dst + c);
  }
}

// init + DIRECT f16 scatter to the per-position sum rows (fast-path variant
// of init_gather_kernel: the all-single-ID path has exactly one id per
// segment, so segment p's sum IS the row of the unique key at sorted
// position p — writing sums[perm[p]] straight from the arena skips the
// [nnz, dim] f32 rows round-trip that segment_sum would re-read).
__global__ void init_scatter_kernel(float* __restrict__ arena,
                                    const ull* __restrict__ query,
                                    const long long* __restrict__ slots,
                                    const int* __restrict__ is_new,
                                    const int64_t* __restrict__ perm,
                                    const int64_t* __restrict__ ustarts,
                                    __half* __restrict__ sums, int64_t n,
                                    int dim, int row_width, double lo,
                                    double hi, float state_init,
                                    const long long* __restrict__ n_dev) {
  if (n_dev) n = *n_dev;
  const int wave = threadIdx.x / PA_WAVE;
  const int lane = threadIdx.x % PA_WAVE;
  const int waves_per_block = blockDim.x / PA_WAVE;
  const int G = (dim < PA_WAVE && (PA_WAVE % dim) == 0) ? PA_WAVE / dim : 1;
  const int sub = (G > 1) ? lane / dim : 0;
  const int c0 = (G > 1) ? lane % dim : lane;
  const int st = (G > 1) ? dim : PA_WAVE;
  const int64_t n_waveitems = (n + G - 1) / G;
  for (int64_t w = (int64_t)blockIdx.x * waves_per_block + wave;
       w < n_waveitems; w += (int64_t)gridDim.x * waves_per_block) {
    const int64_t i = w * G + sub;
    if (i >= n) continue;
    const long long slot = slots[i];
    const int64_t lo_p = ustarts[i], hi_p = ustarts[i + 1];
    if (slot < 0) {
      for (int64_t p = lo_p; p < hi_p; ++p) {
        __half* dst = sums + perm[p] * dim;
        for (int c = c0; c < dim; c += st) dst[c] = __float2half(0.0f);
      }
      continue;
    }
    float* row = arena + (int64_t)slot * row_width;
    if (is_new[i]) {
      const uint64_t sign = pa_splitmix64_inv((uint64_t)query[i]);
      const uint64_t seed = pa_init_seed(sign);
      for (int c = c0; c < dim; c += st)
        row[c] = pa_init_val(seed, c, lo, hi);
      for (int c = dim + c0; c < row_width; c += st)
        row[c] = state_init;
    }
    for (int64_t p = lo_p; p < hi_p; ++p) {
      __half* dst = sums + perm[p] * dim;
      for (int c = c0; c < dim; c += st) dst[c] = __float2half(row[c]);
    }
  }
}

// Dual-key init/scatter for full-wave dims (see scatter_update2_kernel for
// the rationale: the slots->arena->sums chain is latency-bound on random
// row granules; two independent keys per wave double the in-flight loads).
// The rare fresh-init path falls back to per-key sequential work.
__global__ void init_scatter2_kernel(float* __restrict__ arena,
                                     const ull* __restrict__ query,
                                     const long long* __restrict__ slots,
                                     const int* __restrict__ is_new,
                                     const int64_t* __restrict__ perm,
                                     const int64_t* __restrict__ ustarts,
                                     __half* __restrict__ sums, int64_t n,
                                     int dim, int row_width, double lo,
                                     double hi, float state_init,
                                     const long long* __restrict__ n_dev) {
  if (n_dev) n = *n_dev;
  const int wave = threadIdx.x / PA_WAVE;
  const int lane = threadIdx.x % PA_WAVE;
  const int waves_per_block = blockDim.x / PA_WAVE;
  const int64_t n_waveitems = (n + 1) / 2;
  for (int64_t w = (int64_t)blockIdx.x * waves_per_block + wave;
       w < n_waveitems; w += (int64_t)gridDim.x * waves_per_block) {
    const int64_t iA = w * 2, iB = w * 2 + 1;
    const bool okB0 = iB < n;
    const long long slotA = slots[iA];
    const long long slotB = okB0 ? slots[iB] : -1;
    // fresh rows: seeded init first (rare in steady state)
    if (is_new[iA] && slotA >= 0) {
      float* row = arena + (int64_t)slotA * row_width;
      const uint64_t seed = pa_init_seed(pa_splitmix64_inv((uint64_t)query[iA]));
      for (int c = lane; c < dim; c += PA_WAVE) row[c] = pa_init_val(seed, c, lo, hi);
      for (int c = dim + lane; c < row_width; c += PA_WAVE) row[c] = state_init;
    }
    if (okB0 && is_new[iB] && slotB >= 0) {
      float* row = arena + (int64_t)slotB * row_width;
      const uint64_t seed = pa_init_seed(pa_splitmix64_inv((uint64_t)query[iB]));
      for (int c = lane; c < dim; c += PA_WAVE) row[c] = pa_init_val(seed, c, lo, hi);
      for (int c = dim + lane; c < row_width; c += PA_WAVE) row[c] = state_init;
    }
    const int64_t loA = ustarts[iA], hiA = ustarts[iA + 1];
    const int64_t loB = okB0 ? ustarts[iB] : 0;
    const int64_t hiB = okB0 ? ustarts[iB + 1] : 0;
    const float* rowA = slotA >= 0 ? arena + (int64_t)slotA * row_width : nullptr;
    const float* rowB = slotB >= 0 ? arena + (int64_t)slotB * row_width : nullptr;
    if (hiA - loA == 1 && hiB - loB == 1) {  // common single-position case
      __half* dstA = sums + perm[loA] * dim;
      __half* dstB = sums + perm[loB] * dim;
      for (int c = lane; c < dim; c += PA_WAVE) {
        const float vA = rowA ? rowA[c] : 0.0f;
        const float vB = rowB ? rowB[c] : 0.0f;
        dstA[c] = __float2half(vA);
        dstB[c] = __float2half(vB);
      }
    } else {
      for (int64_t p = loA; p < hiA; ++p) {
        __half* dst = sums + perm[p] * dim;
        for (int c = lane; c < dim; c += PA_WAVE)
          dst[c] = __float2half(rowA ? rowA[c] : 0.0f);
      }
      for (int64_t p = loB; p < hiB; ++p) {
        __half* dst = sums + perm[p] * dim;
        for (int c = lane; c < dim; c += PA_WAVE)
          dst[c] = __float2half(rowB ? rowB[c] : 0.0f);
      }
    }
  }
}

// ------------------------------------------------------------- sparse update

// one wave per key: probe (lane-parallel over the 32-slot window) + fused
// row-wise optimizer + weight bound.
// opt: 0 = SGD, 1 = Adagrad, 2 = Adam.   params: see store.py _opt_params.
__global__ void update_kernel(ull* __restrict__ table_keys,
                              unsigned* __restrict__ ticks,
                              float* __restrict__ arena,
                              const ull* __restrict__ query,
                              const float* __restrict__ grads, int64_t n,
                              int dim, int row_width, int64_t n_buckets,
                              int opt, float p0, float p1, float p2, float p3,
                              float b1_power, float b2_power,
                              float weight_bound,
                              int* __restrict__ skipped) {
  const int wave = threadIdx.x / PA_WAVE;
  const int lane = threadIdx.x % PA_WAVE;
  const int waves_per_block = blockDim.x / PA_WAVE;
  const int64_t mask = n_buckets - 1;
  for (int64_t i = (int64_t)blockIdx.x * waves_per_block + wave; i < n;
       i += (int64_t)gridDim.x * waves_per_block) {
    const ull k = query[i];
    if (k == PA_EMPTY_KEY) continue;  // a2a padding: silent skip
    // lane-parallel probe: lanes 0..31 each inspect one slot of the window
    long long slot = -1;
    const int64_t b = (int64_t)(k & (ull)mask);
    if (lane < PA_PROBE_BUCKETS * PA_BUCKET_SIZE) {
      const int p = lane / PA_BUCKET_SIZE;
      const int s = lane % PA_BUCKET_SIZE;
      const int64_t j = ((b + p) & mask) * PA_BUCKET_SIZE + s;
      if (table_keys[j] == k) slot = j;
    }
    const unsigned long long found = __ballot(slot >= 0);
    if (found == 0) {
      if (lane == 0) atomicAdd(&skipped[0], 1);
      continue;
    }
    const int src = __ffsll((long long)found) - 1;
    slot = __shfl(slot, src);
    float* row = arena + (int64_t)slot * row_width;
    const float* g = grads + i * dim;
    // row-level NaN skip (finer-grained than the reference's per-slot skip,
    // mod.rs:731-746: a NaN gradient never touches the table)
    bool has_nan = false;
    for (int c = lane; c < dim; c += PA_WAVE) has_nan |= isnan(g[c]);
    if (__ballot(has_nan) != 0) {
      if (lane == 0) atomicAdd(&skipped[1], 1);
      continue;
    }
    if (opt == 0) {  // SGD: w -= lr*(g + wd*w)         p0=lr p1=wd
      for (int c = lane; c < dim; c += PA_WAVE) {
        float w = row[c] - p0 * (g[c] + p1 * row[c]);
        if (weight_bound > 0.0f) w = fminf(fmaxf(w, -weight_bound), weight_bound);
        row[c] = w;
      }
    } else if (opt == 1) {  // Adagrad  p0=lr p1=g_square_momentum p2=eps p3=vectorwise
      if (p3 > 0.5f) {
        // shared scalar accumulator at row[dim]
        const float acc = row[dim];
        float gsq = 0.0f;
        for (int c = lane; c < dim; c += PA_WAVE) {
          float w = row[c] - p0 * g[c] * rsqrtf(acc + p2);
          if (weight_bound > 0.0f) w = fminf(fmaxf(w, -weight_bound), weight_bound);
          row[c] = w;
          gsq += g[c] * g[c];
        }
#pragma unroll
        for (int off = PA_WAVE / 2; off > 0; off >>= 1)
          gsq += __shfl_down(gsq, off);
        if (lane == 0) row[dim] = acc * p1 + gsq / (float)dim;
      } else {
        for (int c = lane; c < dim; c += PA_WAVE) {
          const float acc = row[dim + c];
          float w = row[c] - p0 * g[c] * rsqrtf(acc + p2);
          if (weight_bound > 0.0f) w = fminf(fmaxf(w, -weight_bound), weight_bound);
          row[c] = w;
          row[dim + c] = acc * p1 + g[c] * g[c];
        }
      }
    } else {  // Adam  p0=lr p1=beta1 p2=beta2 p3=eps
      const float om1 = 1.0f - p1, om2 = 1.0f - p2;
      const float c1 = 1.0f / (1.0f - b1_power), c2 = 1.0f / (1.0f - b2_power);
      for (int c = lane; c < dim; c += PA_WAVE) {
        const float m = p1 * row[dim + c] + om1 * g[c];
        const float v = p2 * row[2 * dim + c] + om2 * g[c] * g[c];
        float w = row[c] - p0 * (m * c1) / (p3 + sqrtf(v * c2));
        if (weight_bound > 0.0f) w = fminf(fmaxf(w, -weight_bound), weight_bound);
        row[c] = w;
        row[dim + c] = m;
        row[2 * dim + c] = v;
      }
    }
  }
}

// ------------------------------------------------------------------ sign prep

// raw ids -> routed keys: fold into the slot's feature-group prefix space
// (indices_add_prefix, mod.rs:403-429) then splitmix64-mix (the mixed key is
// the dedup sort key AND the shard router).  Slot of position i found by
// binary search over slot_starts (S is small).
__global__ void sign_prep_kernel(const ull* __restrict__ values,
                                 const int64_t* __restrict__ slot_starts,
                                 const ull* __restrict__ prefixes, int n_slots,
                                 ull spacing, ull* __restrict__ out,
                                 int64_t n) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  int lo = 0, hi = n_slots;  // find slot: slot_starts[s] <= i < slot_starts[s+1]
  while (hi - lo > 1) {
    const int mid = (lo + hi) >> 1;
    if (i >= slot_starts[mid]) lo = mid; else hi = mid;
  }
  const ull prefix = prefixes[lo];
  ull sign = values[i];
  if (prefix != 0) sign = sign % spacing + prefix;
  ull k = pa_splitmix64(sign);
  if (k == PA_EMPTY_KEY) k = 0xD1B54A32D192ED03ull;  // empty-sentinel remap
  out[i] = k;
}

// sign prep with hashstack expansion (reference
// indices_to_hashstack_indices, embedding_worker_service/mod.rs:348-400):
// each raw id of a slot with hash_stack_rounds=R becomes R bucketed ids
// (round r: iterate splitmix64, fold into [r*size,(r+1)*size)), laid out
// position-major per sample ([r0 ids..., r1 ids...] inside each sample's
// span) to match the CPU path bit-for-bit.  One thread per INPUT id writes
// its R outputs; slots with R==0 pass through the plain prefix+mix path.
__global__ void sign_prep_stack_kernel(
    const ull* __restrict__ values, const int64_t* __restrict__ in_starts,
    const int64_t* __restrict__ out_starts, const ull* __restrict__ prefixes,
    const int* __restrict__ rounds, const int64_t* __restrict__ sizes,
    const int64_t* __restrict__ off_starts,
    const int64_t* __restrict__ all_offs,  // EXPANDED per-slot seg offsets
    int n_slots, ull spacing, ull* __restrict__ out, int64_t n_in) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n_in) return;
  int lo = 0, hi = n_slots;  // find slot: in_starts[s] <= i < in_starts[s+1]
  while (hi - lo > 1) {
    const int mid = (lo + hi) >> 1;
    if (i >= in_starts[mid]) lo = mid; else hi = mid;
  }
  const int s = lo;
  const ull prefix = prefixes[s];
  const int R = rounds[s];
  if (R == 0) {
    ull sign = values[i];
    if (prefix != 0) sign = sign % spacing + prefix;
    ull k = pa_splitmix64(sign);
    if (k == PA_EMPTY_KEY) k = 0xD1B54A32D192ED03ull;
    out[out_starts[s] + (i - in_starts[s])] = k;
    return;
  }
  const int64_t j = i - in_starts[s];  // slot-local input position
  // sample b: expanded offsets E[b] = off[b]*R, so E[b] <= j*R < E[b+1]
  const int64_t* E = all_offs + off_starts[s];
  const int64_t nE = off_starts[s + 1] - off_starts[s];
  const int64_t key = j * (int64_t)R;
  int64_t blo = 0, bhi = nE - 1;
  while (bhi - blo > 1) {
    const int64_t mid = (blo + bhi) >> 1;
    if (key >= E[mid]) blo = mid; else bhi = mid;
  }
  const int64_t seg_lo = E[blo] / R, seg_hi = E[blo + 1] / R;
  const int64_t base = out_starts[s] + E[blo];
  const int64_t size = sizes[s];
  ull h = values[i];
  for (int r = 0; r < R; ++r) {
    h = pa_splitmix64(h);
    ull sign = h % (ull)size + (ull)((int64_t)r * size);
    if (prefix != 0) sign = sign % spacing + prefix;
    ull k = pa_splitmix64(sign);
    if (k == PA_EMPTY_KEY) k = 0xD1B54A32D192ED03ull;
    out[base + (int64_t)r * (seg_hi - seg_lo) + (j - seg_lo)] = k;
  }
}

// ------------------------------------------------------- padded a2a routing
// The owner partition of the SORTED uniq keys is a range partition (owner =
// (hi32*world)>>32 is monotone in the key), so per-owner segment starts are
// world binary searches — one tiny kernel — and the pack is one grid-stride
// scatter.  Replaces a ~10-dispatch torch chain that cost the producer
// thread ~0.5 ms/batch of issue time.
__global__ void a2a_bounds_kernel(const ull* __restrict__ uniq,
                                  const long long* __restrict__ n_dev,
                                  long long n_pad, int world,
                                  long long* __restrict__ starts) {
  const long long n = n_dev ? *n_dev : n_pad;
  const int r = threadIdx.x;
  if (r > world) return;
  if (r == 0) { starts[0] = 0; return; }
  if (r == world) { starts[world] = n; return; }
  // smallest key with owner >= r: hi32 >= ceil(r * 2^32 / world)
  const ull hi_min =
      (ull)((((unsigned __int128)r << 32) + (ull)world - 1) / (ull)world);
  const ull thresh = hi_min << 32;
  long long lo = 0, hi = n;
  while (lo < hi) {
    const long long mid = (lo + hi) >> 1;
    if (uniq[mid] < thresh) lo = mid + 1; else hi = mid;
  }
  starts[r] = lo;
}

__global__ void a2a_scatter_kernel(const ull* __restrict__ uniq,
                                   const long long* __restrict__ n_dev,
                                   long long n_pad,
                                   const long long* __restrict__ starts,
                                   int world, long long cap,
                                   ull* __restrict__ send,
                                   int64_t* __restrict__ idx,
                                   long long* __restrict__ overflow) {
  const long long n = n_dev ? *n_dev : n_pad;
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n_pad) return;
  const long long dummy = (long long)world * cap;
  if (i >= n) {  // dedup padding tail
    idx[i] = dummy;
    return;
  }
  const ull k = uniq[i];
  const int o = (int)(((k >> 32) * (ull)world) >> 32);
  const long long pos = i - starts[o];
  long long d = dummy;
  if (pos < cap) {
    d = (long long)o * cap + pos;
    send[d] = k;
  } else {
    atomicAdd((unsigned long long*)overflow, 1ull);
  }
  idx[i] = d;
}

__global__ void iota_i64_kernel(int64_t* __restrict__ out, int64_t n) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) out[i] = i;
}

// flags[i] = (sorted[i] != sorted[i-1]); flags[0] = 1  (boundary marks for
// the rank scan — replaces a ones+slice+ne op chain)
__global__ void neq_flags_kernel(const ull* __restrict__ sorted,
                                 bool* __restrict__ flags, int64_t n) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  flags[i] = (i == 0) || (sorted[i] != sorted[i - 1]);
}

// Fixed-shape dedup epilogue: after sort + neq + rank=cumsum(neq)-1 (all
// shape-static ATen ops), ONE pass finalizes inverse/uniq/ustarts into
// nnz-padded buffers and writes the true unique count to a device scalar —
// the host never learns U, so the whole lookup issues without a single
// synchronization (the probe/gather/update kernels read U from n_dev).
__global__ void dedup_finalize_kernel(
    const ull* __restrict__ svals_flipped, const int64_t* __restrict__ perm,
    const bool* __restrict__ neq, const int64_t* __restrict__ rank,
    int64_t n, ull flip, int64_t* __restrict__ inverse,
    ull* __restrict__ uniq, int64_t* __restrict__ ustarts,
    long long* __restrict__ u_count) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const int64_t r = rank[i];
  inverse[perm[i]] = r;
  if (neq[i]) {
    uniq[r] = svals_flipped[i] ^ flip;
    ustarts[r] = i;
  }
  if (i == n - 1) {
    u_count[0] = r + 1;
    ustarts[r + 1] = n;
  }
}

// --------------------------------------------------------------- import rows

__global__ void import_kernel(ull* __restrict__ table_keys,
                              unsigned* __restrict__ ticks,
                              float* __restrict__ arena,
                              const ull* __restrict__ query,
                              const float* __restrict__ rows, int64_t n,
                              int row_width, int64_t n_buckets, unsigned tick,
                              ull* __restrict__ evict_keys,
                              int* __restrict__ evict_count,
                              float* __restrict__ evict_rows) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  int is_new;
  ull victim = 0;
  const long long slot =
      probe_claim(table_keys, ticks, n_buckets, query[i], tick, &is_new,
                  evict_keys ? &victim : nullptr);
  if (slot < 0) return;
  ticks[slot] = tick;
  float* dst = arena + (int64_t)slot * row_width;
  if (evict_keys && victim != 0) {
    const long long eidx = (long long)atomicAdd(evict_count, 1);
    evict_keys[eidx] = victim;
    float* spill = evict_rows + eidx * row_width;
    for (int c = 0; c < row_width; ++c) spill[c] = dst[c];
  }
  const float* src = rows + i * row_width;
  for (int c = 0; c < row_width; ++c) dst[c] = src[c];
}

// ----------------------------------------------------- fused gather + seg-sum

// out[s, :] = seg_scale[s] * sum_{k in [off[s], off[s+1])} rows[inverse[k], :]
// One wave per segment (all sum-slots of a dim-group in ONE launch);
// f32 accumulate; f16 store (reference wire dtype, persia-common lib.rs:88-99).
// Small dims pack 64/dim segments per wave (sub-wave groups) so dim-8
// tables still use all lanes; large dims stride as before.
template <typename RowT>
__global__ void segment_sum_kernel(const RowT* __restrict__ rows,
                                   const int64_t* __restrict__ inverse,
                                   const int64_t* __restrict__ seg_offsets,
                                   const float* __restrict__ seg_scale,
                                   __half* __restrict__ out, int64_t n_seg,
                                   int dim) {
  const int wave = threadIdx.x / PA_WAVE;
  const int lane = threadIdx.x % PA_WAVE;
  const int waves_per_block = blockDim.x / PA_WAVE;
  const int G = (dim < PA_WAVE && (PA_WAVE % dim) == 0) ? PA_WAVE / dim : 1;
  const int sub = (G > 1) ? lane / dim : 0;
  const int c0 = (G > 1) ? lane % dim : lane;
  const int64_t n_waveitems = (n_seg + G - 1) / G;
  for (int64_t w = (int64_t)blockIdx.x * waves_per_block + wave;
       w < n_waveitems; w += (int64_t)gridDim.x * waves_per_block) {
    const int64_t s = w * G + sub;
    if (s >= n_seg) continue;
    const int64_t lo = seg_offsets[s], hi = seg_offsets[s + 1];
    const float scale = seg_scale ? seg_scale[s] : 1.0f;
    __half* dst = out + s * dim;
    for (int c = c0; c < dim; c += PA_WAVE) {
      float acc = 0.0f;
      for (int64_t k = lo; k < hi; ++k) {
        acc += pa_to_float(rows[inverse[k] * dim + c]);
      }
      dst[c] = __float2half(acc * scale);
    }
  }
}

// --------------------------------------------------- ordered gradient scatter

// out[u, :] += sum_{p in [ustart[u], ustart[u+1])} grads[seg_id[perm[p]], :]
//              * seg_scale[seg_id[perm[p]]]
// One wave per unique sign — deterministic (ordered), atomic-free: the
// positions of a unique key are contiguous in sort order.  Positions with
// seg_id < 0 (raw-slot positions, handled by a separate indexed add) are
// skipped.
template <typename GradT, bool ACCUMULATE>
__global__ void grad_scatter_kernel(const GradT* __restrict__ grads,
                                    const int64_t* __restrict__ perm,
                                    const int64_t* __restrict__ ustarts,
                                    const int64_t* __restrict__ seg_id,
                                    const float* __restrict__ seg_scale,
                                    float* __restrict__ out, int64_t n_unique,
                                    int dim) {
  const int wave = threadIdx.x / PA_WAVE;
  const int lane = threadIdx.x % PA_WAVE;
  const int waves_per_block = blockDim.x / PA_WAVE;
  for (int64_t u = (int64_t)blockIdx.x * waves_per_block + wave; u < n_unique;
       u += (int64_t)gridDim.x * waves_per_block) {
    const int64_t lo = ustarts[u], hi = ustarts[u + 1];
    float* dst = out + u * dim;
    for (int c = lane; c < dim; c += PA_WAVE) {
      float acc = 0.0f;
      for (int64_t p = lo; p < hi; ++p) {
        const int64_t s = seg_id[perm[p]];
        if (s >= 0) {
          // scale 0 skips the slot entirely (NaN-slot mask: never multiply a
          // NaN gradient, even by zero); null scale = 1
          const float sc = seg_scale ? seg_scale[s] : 1.0f;
          if (sc != 0.0f) acc += pa_to_float(grads[s * dim + c]) * sc;
        }
      }
      if (ACCUMULATE) dst[c] += acc; else dst[c] = acc;
    }
  }
}

// Indexed wire variant: out[out_idx[u], :] = acc in the WIRE dtype — packs
// each unique key's reduced gradient straight into the a2a send buffer
// (layout [world, cap] by owner rank), skipping both the [U, dim] f32
// intermediate and a separate pack pass.  n_unique comes from n_dev (padded
// sync-free dedup); invalid/overflowed entries point at the dummy tail row.
// Small dims pack 64/dim keys per wave (sub-wave groups).
template <typename OutT>
__global__ void grad_scatter_idx_kernel(
    const __half* __restrict__ grads, const int64_t* __restrict__ perm,
    const int64_t* __restrict__ ustarts, const int64_t* __restrict__ seg_id,
    const float* __restrict__ seg_scale, const int64_t* __restrict__ out_idx,
    OutT* __restrict__ out, int64_t n_unique, int dim,
    const long long* __restrict__ n_dev) {
  if (n_dev) n_unique = *n_dev;
  const int wave = threadIdx.x / PA_WAVE;
  const int lane = threadIdx.x % PA_WAVE;
  const int waves_per_block = blockDim.x / PA_WAVE;
  const int G = (dim < PA_WAVE && (PA_WAVE % dim) == 0) ? PA_WAVE / dim : 1;
  const int sub = (G > 1) ? lane / dim : 0;
  const int c0 = (G > 1) ? lane % dim : lane;
  const int st = (G > 1) ? dim : PA_WAVE;
  const int64_t n_waveitems = (n_unique + G - 1) / G;
  for (int64_t w = (int64_t)blockIdx.x * waves_per_block + wave;
       w < n_waveitems; w += (int64_t)gridDim.x * waves_per_block) {
    const int64_t u = w * G + sub;
    if (u >= n_unique) continue;
    const int64_t lo = ustarts[u], hi = ustarts[u + 1];
    OutT* dst = out + out_idx[u] * dim;
    for (int c = c0; c < dim; c += st) {
      float acc = 0.0f;
      for (int64_t p = lo; p < hi; ++p) {
        const int64_t s = seg_id[perm[p]];
        if (s >= 0) {
          const float sc = seg_scale ? seg_scale[s] : 1.0f;
          if (sc != 0.0f) acc += __half2float(grads[s * dim + c]) * sc;
        }
      }
      pa_store_out(acc, dst + c);
    }
  }
}

// ------------------------------------------- fused scatter + optimizer update

// One wave per unique sign: reduce its gradient from the per-segment grads
// (ordered, via the sort permutation) straight into registers, probe the
// table, and apply the optimizer — no intermediate [U, dim] buffer.
// Supports dim <= 8*PA_WAVE (register-resident grad row).
__global__ void scatter_update_kernel(
    ull* __restrict__ table_keys, unsigned* __restrict__ ticks,
    float* __restrict__ arena, const ull* __restrict__ uniq,
    const __half* __restrict__ grads, const int64_t* __restrict__ perm,
    const int64_t* __restrict__ ustarts, const int64_t* __restrict__ seg_id,
    const float* __restrict__ seg_scale, int64_t n, int dim, int row_width,
    int64_t n_buckets, int opt, float p0, float p1, float p2, float p3,
    float b1_power, float b2_power, float weight_bound,
    int* __restrict__ skipped, const long long* __restrict__ n_dev) {
  if (n_dev) n = *n_dev;
  const int wave = threadIdx.x / PA_WAVE;
  const int lane = threadIdx.x % PA_WAVE;
  const int waves_per_block = blockDim.x / PA_WAVE;
  const int64_t mask = n_buckets - 1;
  // small dims pack 64/dim keys per wave (sub-wave groups)
  const int G = (dim < PA_WAVE && (PA_WAVE % dim) == 0) ? PA_WAVE / dim : 1;
  const int sub = (G > 1) ? lane / dim : 0;
  const int c0 = (G > 1) ? lane % dim : lane;
  const int st = (G > 1) ? dim : PA_WAVE;
  const int sg = (G > 1) ? dim : PA_WAVE;  // subgroup width
  const ull sg_mask = (sg >= 64) ? ~0ull : ((1ull << sg) - 1ull);
  const int64_t n_waveitems = (n + G - 1) / G;
  constexpr int WINDOW = PA_PROBE_BUCKETS * PA_BUCKET_SIZE;
  for (int64_t w = (int64_t)blockIdx.x * waves_per_block + wave;
       w < n_waveitems; w += (int64_t)gridDim.x * waves_per_block) {
    const int64_t u = w * G + sub;
    const ull k = (u < n) ? uniq[u] : 0;
    // empty key = a2a padding: silent skip (the empty key would match every
    // free table slot in the probe)
    const bool active = (u < n) && (k != PA_EMPTY_KEY);
    // subgroup-parallel probe over the 32-slot window
    long long slot = -1;
    if (active) {
      const int64_t b = (int64_t)(k & (ull)mask);
      for (int it = 0; it * sg + c0 < WINDOW; ++it) {
        const int widx = it * sg + c0;
        const int p = widx / PA_BUCKET_SIZE;
        const int s = widx % PA_BUCKET_SIZE;
        const int64_t j = ((b + p) & mask) * PA_BUCKET_SIZE + s;
        if (table_keys[j] == k) slot = j;
      }
    }
    const unsigned long long found = __ballot(slot >= 0);
    const unsigned long long sub_found = (found >> (sub * sg)) & sg_mask;
    bool ok = active;
    if (ok && sub_found == 0) {
      if (c0 == 0) atomicAdd(&skipped[0], 1);
      ok = false;
    }
    if (ok)
      slot = __shfl(slot, sub * sg + __ffsll((long long)sub_found) - 1);
    // ordered gradient reduction into registers
    float acc[8];
#pragma unroll
    for (int t = 0; t < 8; ++t) acc[t] = 0.0f;
    if (ok) {
      const int64_t lo = ustarts[u], hi = ustarts[u + 1];
      for (int64_t p = lo; p < hi; ++p) {
        const int64_t s = seg_id[perm[p]];
        if (s < 0) continue;
        const float sc = seg_scale ? seg_scale[s] : 1.0f;
        if (sc == 0.0f) continue;
        const __half* g = grads + s * dim;
        for (int t = 0, c = c0; c < dim; c += st, ++t)
          acc[t] += __half2float(g[c]) * sc;
      }
    }
    bool has_nan = false;
    if (ok)
      for (int t = 0, c = c0; c < dim; c += st, ++t) has_nan |= isnan(acc[t]);
    const unsigned long long nanb = __ballot(has_nan);
    if (ok && ((nanb >> (sub * sg)) & sg_mask) != 0) {
      if (c0 == 0) atomicAdd(&skipped[1], 1);
      ok = false;
    }
    if (!ok) continue;
    float* row = arena + (int64_t)slot * row_width;
    // nontemporal arena traffic: each row is touched exactly once per step
    // and this kernel overlaps the dense graph on another stream — keeping
    // ~0.5 GB/step of one-shot data out of L2 leaves it to the GEMMs
    if (opt == 0) {  // SGD
      for (int t = 0, c = c0; c < dim; c += st, ++t) {
        const float w0 = __builtin_nontemporal_load(&row[c]);
        float w2 = w0 - p0 * (acc[t] + p1 * w0);
        if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
        __builtin_nontemporal_store(w2, &row[c]);
      }
    } else if (opt == 1) {  // Adagrad
      if (p3 > 0.5f) {
        const float a0 = row[dim];
        float gsq = 0.0f;
        for (int t = 0, c = c0; c < dim; c += st, ++t) {
          const float w0 = __builtin_nontemporal_load(&row[c]);
          float w2 = w0 - p0 * acc[t] * rsqrtf(a0 + p2);
          if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
          __builtin_nontemporal_store(w2, &row[c]);
          gsq += acc[t] * acc[t];
        }
        for (int off = sg / 2; off > 0; off >>= 1)
          gsq += __shfl_down(gsq, off, sg);
        if (c0 == 0) row[dim] = a0 * p1 + gsq / (float)dim;
      } else {
        for (int t = 0, c = c0; c < dim; c += st, ++t) {
          const float a0 = __builtin_nontemporal_load(&row[dim + c]);
          const float w0 = __builtin_nontemporal_load(&row[c]);
          float w2 = w0 - p0 * acc[t] * rsqrtf(a0 + p2);
          if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
          __builtin_nontemporal_store(w2, &row[c]);
          __builtin_nontemporal_store(a0 * p1 + acc[t] * acc[t], &row[dim + c]);
        }
      }
    } else {  // Adam
      const float om1 = 1.0f - p1, om2 = 1.0f - p2;
      const float c1 = 1.0f / (1.0f - b1_power), c2 = 1.0f / (1.0f - b2_power);
      for (int t = 0, c = c0; c < dim; c += st, ++t) {
        const float m = p1 * __builtin_nontemporal_load(&row[dim + c]) + om1 * acc[t];
        const float v =
            p2 * __builtin_nontemporal_load(&row[2 * dim + c]) + om2 * acc[t] * acc[t];
        const float w0 = __builtin_nontemporal_load(&row[c]);
        float w2 = w0 - p0 * (m * c1) / (p3 + sqrtf(v * c2));
        if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
        __builtin_nontemporal_store(w2, &row[c]);
        __builtin_nontemporal_store(m, &row[dim + c]);
        __builtin_nontemporal_store(v, &row[2 * dim + c]);
      }
    }
  }
}

// Dual-key variant for full-wave dims (64 <= dim <= 256): one wave updates
// TWO unique keys with interleaved loads — the single-key kernel is
// latency-bound on its probe -> grads -> arena dependent chain (random 1-2 KB
// row granules), so doubling the independent in-flight accesses per wave
// buys memory-level parallelism the scheduler can't extract across
// grid-stride iterations (ballots/shfl block software pipelining).
__global__ void scatter_update2_kernel(
    ull* __restrict__ table_keys, unsigned* __restrict__ ticks,
    float* __restrict__ arena, const ull* __restrict__ uniq,
    const __half* __restrict__ grads, const int64_t* __restrict__ perm,
    const int64_t* __restrict__ ustarts, const int64_t* __restrict__ seg_id,
    const float* __restrict__ seg_scale, int64_t n, int dim, int row_width,
    int64_t n_buckets, int opt, float p0, float p1, float p2, float p3,
    float b1_power, float b2_power, float weight_bound,
    int* __restrict__ skipped, const long long* __restrict__ n_dev) {
  if (n_dev) n = *n_dev;
  const int wave = threadIdx.x / PA_WAVE;
  const int lane = threadIdx.x % PA_WAVE;
  const int waves_per_block = blockDim.x / PA_WAVE;
  const int64_t mask = n_buckets - 1;
  constexpr int WINDOW = PA_PROBE_BUCKETS * PA_BUCKET_SIZE;  // 32
  const int half = lane >> 5;   // probe phase: lanes 0-31 key A, 32-63 key B
  const int widx = lane & 31;
  const int64_t n_waveitems = (n + 1) / 2;
  for (int64_t w = (int64_t)blockIdx.x * waves_per_block + wave;
       w < n_waveitems; w += (int64_t)gridDim.x * waves_per_block) {
    const int64_t u = w * 2 + half;
    const ull k = (u < n) ? uniq[u] : 0;
    const bool active = (u < n) && (k != PA_EMPTY_KEY);
    long long slot = -1;
    if (active) {
      const int64_t b = (int64_t)(k & (ull)mask);
      const int p = widx / PA_BUCKET_SIZE;
      const int s = widx % PA_BUCKET_SIZE;
      const int64_t j = ((b + p) & mask) * PA_BUCKET_SIZE + s;
      if (table_keys[j] == k) slot = j;
    }
    const unsigned long long found = __ballot(slot >= 0);
    const unsigned long long nonzero = __ballot(k != PA_EMPTY_KEY);
    const unsigned fA = (unsigned)(found & 0xFFFFFFFFull);
    const unsigned fB = (unsigned)(found >> 32);
    // empty-key (a2a padding) entries skip silently, not as misses
    bool okA = (w * 2 + 0) < n && (nonzero & 1ull);
    bool okB = (w * 2 + 1) < n && (nonzero >> 32 & 1ull);
    if (okA && fA == 0) { if (lane == 0) atomicAdd(&skipped[0], 1); okA = false; }
    if (okB && fB == 0) { if (lane == 0) atomicAdd(&skipped[0], 1); okB = false; }
    const long long slotA = okA ? __shfl(slot, __ffs(fA) - 1) : -1;
    const long long slotB = okB ? __shfl(slot, 32 + __ffs(fB) - 1) : -1;
    // gradient accumulation (whole wave per key, interleaved when both are
    // single-position — the overwhelmingly common case)
    float accA[4] = {}, accB[4] = {};
    const int64_t loA = ustarts[w * 2], hiA = okA ? ustarts[w * 2 + 1] : loA;
    const int64_t loB = okB ? ustarts[w * 2 + 1] : 0;
    const int64_t hiB = okB ? ustarts[w * 2 + 2] : loB;
    if (okA && okB && hiA - loA == 1 && hiB - loB == 1) {
      const int64_t sA = seg_id[perm[loA]], sB = seg_id[perm[loB]];
      const float scA = (sA >= 0) ? (seg_scale ? seg_scale[sA] : 1.0f) : 0.0f;
      const float scB = (sB >= 0) ? (seg_scale ? seg_scale[sB] : 1.0f) : 0.0f;
      const __half* gA = grads + (sA >= 0 ? sA : 0) * dim;
      const __half* gB = grads + (sB >= 0 ? sB : 0) * dim;
      for (int t = 0, c = lane; c < dim; c += PA_WAVE, ++t) {
        accA[t] = scA * __half2float(gA[c]);
        accB[t] = scB * __half2float(gB[c]);
      }
    } else {
      if (okA)
        for (int64_t p = loA; p < hiA; ++p) {
          const int64_t s = seg_id[perm[p]];
          if (s < 0) continue;
          const float sc = seg_scale ? seg_scale[s] : 1.0f;
          if (sc == 0.0f) continue;
          const __half* g = grads + s * dim;
          for (int t = 0, c = lane; c < dim; c += PA_WAVE, ++t)
            accA[t] += __half2float(g[c]) * sc;
        }
      if (okB)
        for (int64_t p = loB; p < hiB; ++p) {
          const int64_t s = seg_id[perm[p]];
          if (s < 0) continue;
          const float sc = seg_scale ? seg_scale[s] : 1.0f;
          if (sc == 0.0f) continue;
          const __half* g = grads + s * dim;
          for (int t = 0, c = lane; c < dim; c += PA_WAVE, ++t)
            accB[t] += __half2float(g[c]) * sc;
        }
    }
    bool nanA = false, nanB = false;
    for (int t = 0, c = lane; c < dim; c += PA_WAVE, ++t) {
      nanA |= isnan(accA[t]);
      nanB |= isnan(accB[t]);
    }
    if (okA && __ballot(nanA) != 0) {
      if (lane == 0) atomicAdd(&skipped[1], 1);
      okA = false;
    }
    if (okB && __ballot(nanB) != 0) {
      if (lane == 0) atomicAdd(&skipped[1], 1);
      okB = false;
    }
    if (!okA && !okB) continue;
    float* rowA = okA ? arena + (int64_t)slotA * row_width : nullptr;
    float* rowB = okB ? arena + (int64_t)slotB * row_width : nullptr;
    if (opt == 0) {  // SGD
      for (int t = 0, c = lane; c < dim; c += PA_WAVE, ++t) {
        const float wA = okA ? __builtin_nontemporal_load(&rowA[c]) : 0.0f;
        const float wB = okB ? __builtin_nontemporal_load(&rowB[c]) : 0.0f;
        if (okA) {
          float w2 = wA - p0 * (accA[t] + p1 * wA);
          if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
          __builtin_nontemporal_store(w2, &rowA[c]);
        }
        if (okB) {
          float w2 = wB - p0 * (accB[t] + p1 * wB);
          if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
          __builtin_nontemporal_store(w2, &rowB[c]);
        }
      }
    } else if (opt == 1) {  // Adagrad
      if (p3 > 0.5f) {  // vectorwise-shared accumulator
        const float a0A = okA ? rowA[dim] : 0.0f;
        const float a0B = okB ? rowB[dim] : 0.0f;
        float gsqA = 0.0f, gsqB = 0.0f;
        for (int t = 0, c = lane; c < dim; c += PA_WAVE, ++t) {
          const float wA = okA ? __builtin_nontemporal_load(&rowA[c]) : 0.0f;
          const float wB = okB ? __builtin_nontemporal_load(&rowB[c]) : 0.0f;
          if (okA) {
            float w2 = wA - p0 * accA[t] * rsqrtf(a0A + p2);
            if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
            __builtin_nontemporal_store(w2, &rowA[c]);
            gsqA += accA[t] * accA[t];
          }
          if (okB) {
            float w2 = wB - p0 * accB[t] * rsqrtf(a0B + p2);
            if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
            __builtin_nontemporal_store(w2, &rowB[c]);
            gsqB += accB[t] * accB[t];
          }
        }
#pragma unroll
        for (int off = PA_WAVE / 2; off > 0; off >>= 1) {
          gsqA += __shfl_down(gsqA, off);
          gsqB += __shfl_down(gsqB, off);
        }
        if (lane == 0) {
          if (okA) rowA[dim] = a0A * p1 + gsqA / (float)dim;
          if (okB) rowB[dim] = a0B * p1 + gsqB / (float)dim;
        }
      } else {
        for (int t = 0, c = lane; c < dim; c += PA_WAVE, ++t) {
          const float aA = okA ? __builtin_nontemporal_load(&rowA[dim + c]) : 0.0f;
          const float aB = okB ? __builtin_nontemporal_load(&rowB[dim + c]) : 0.0f;
          const float wA = okA ? __builtin_nontemporal_load(&rowA[c]) : 0.0f;
          const float wB = okB ? __builtin_nontemporal_load(&rowB[c]) : 0.0f;
          if (okA) {
            float w2 = wA - p0 * accA[t] * rsqrtf(aA + p2);
            if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
            __builtin_nontemporal_store(w2, &rowA[c]);
            __builtin_nontemporal_store(aA * p1 + accA[t] * accA[t], &rowA[dim + c]);
          }
          if (okB) {
            float w2 = wB - p0 * accB[t] * rsqrtf(aB + p2);
            if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
            __builtin_nontemporal_store(w2, &rowB[c]);
            __builtin_nontemporal_store(aB * p1 + accB[t] * accB[t], &rowB[dim + c]);
          }
        }
      }
    } else {  // Adam
      const float om1 = 1.0f - p1, om2 = 1.0f - p2;
      const float c1 = 1.0f / (1.0f - b1_power), c2 = 1.0f / (1.0f - b2_power);
      for (int t = 0, c = lane; c < dim; c += PA_WAVE, ++t) {
        if (okA) {
          const float m = p1 * __builtin_nontemporal_load(&rowA[dim + c]) + om1 * accA[t];
          const float v =
              p2 * __builtin_nontemporal_load(&rowA[2 * dim + c]) + om2 * accA[t] * accA[t];
          const float w0 = __builtin_nontemporal_load(&rowA[c]);
          float w2 = w0 - p0 * (m * c1) / (p3 + sqrtf(v * c2));
          if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
          __builtin_nontemporal_store(w2, &rowA[c]);
          __builtin_nontemporal_store(m, &rowA[dim + c]);
          __builtin_nontemporal_store(v, &rowA[2 * dim + c]);
        }
        if (okB) {
          const float m = p1 * __builtin_nontemporal_load(&rowB[dim + c]) + om1 * accB[t];
          const float v =
              p2 * __builtin_nontemporal_load(&rowB[2 * dim + c]) + om2 * accB[t] * accB[t];
          const float w0 = __builtin_nontemporal_load(&rowB[c]);
          float w2 = w0 - p0 * (m * c1) / (p3 + sqrtf(v * c2));
          if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
          __builtin_nontemporal_store(w2, &rowB[c]);
          __builtin_nontemporal_store(m, &rowB[dim + c]);
          __builtin_nontemporal_store(v, &rowB[2 * dim + c]);
        }
      }
    }
  }
}

// Quad-key variant for dim <= 128 (acc fits 2 cols/lane/key): 4 independent
// probe->grads->arena chains per wave.  Probe: 16 lanes per key, each lane
// scans 2 of the 32 window slots.
__global__ void scatter_update4_kernel(
    ull* __restrict__ table_keys, unsigned* __restrict__ ticks,
    float* __restrict__ arena, const ull* __restrict__ uniq,
    const __half* __restrict__ grads, const int64_t* __restrict__ perm,
    const int64_t* __restrict__ ustarts, const int64_t* __restrict__ seg_id,
    const float* __restrict__ seg_scale, int64_t n, int dim, int row_width,
    int64_t n_buckets, int opt, float p0, float p1, float p2, float p3,
    float b1_power, float b2_power, float weight_bound,
    int* __restrict__ skipped, const long long* __restrict__ n_dev) {
  if (n_dev) n = *n_dev;
  const int wave = threadIdx.x / PA_WAVE;
  const int lane = threadIdx.x % PA_WAVE;
  const int waves_per_block = blockDim.x / PA_WAVE;
  const int64_t mask = n_buckets - 1;
  const int kq = lane >> 4;   // probe phase: 16 lanes per key
  const int widx = lane & 15;
  const int64_t n_waveitems = (n + 3) / 4;
  for (int64_t w = (int64_t)blockIdx.x * waves_per_block + wave;
       w < n_waveitems; w += (int64_t)gridDim.x * waves_per_block) {
    const int64_t u = w * 4 + kq;
    const ull k = (u < n) ? uniq[u] : 0;
    const bool active = (u < n) && (k != PA_EMPTY_KEY);
    long long slot = -1;
    if (active) {
      const int64_t b = (int64_t)(k & (ull)mask);
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        const int wi = widx + half * 16;
        const int p = wi / PA_BUCKET_SIZE;
        const int s = wi % PA_BUCKET_SIZE;
        const int64_t j = ((b + p) & mask) * PA_BUCKET_SIZE + s;
        if (table_keys[j] == k) slot = j;
      }
    }
    const unsigned long long found = __ballot(slot >= 0);
    const unsigned long long nonzero = __ballot(k != PA_EMPTY_KEY);
    bool ok[4];
    long long slotk[4];
    float* rowk[4];
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      const unsigned fq = (unsigned)((found >> (16 * q)) & 0xFFFFull);
      // empty-key (a2a padding) entries skip silently, not as misses
      ok[q] = (w * 4 + q) < n && (nonzero >> (16 * q) & 1ull);
      if (ok[q] && fq == 0) {
        if (lane == 0) atomicAdd(&skipped[0], 1);
        ok[q] = false;
      }
      slotk[q] = ok[q] ? __shfl(slot, 16 * q + __ffs(fq) - 1) : -1;
      rowk[q] = ok[q] ? arena + (int64_t)slotk[q] * row_width : nullptr;
    }
    // gradient accumulation (whole wave per key; interleaved single-position
    // fast path is the overwhelmingly common case)
    float acc[4][2] = {};
    int64_t lop[4], hip[4];
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      lop[q] = ok[q] ? ustarts[w * 4 + q] : 0;
      hip[q] = ok[q] ? ustarts[w * 4 + q + 1] : 0;
    }
    const bool all_single = ok[0] && ok[1] && ok[2] && ok[3] &&
                            hip[0] - lop[0] == 1 && hip[1] - lop[1] == 1 &&
                            hip[2] - lop[2] == 1 && hip[3] - lop[3] == 1;
    if (all_single) {
      const __half* gq[4];
      float scq[4];
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const int64_t s = seg_id[perm[lop[q]]];
        scq[q] = (s >= 0) ? (seg_scale ? seg_scale[s] : 1.0f) : 0.0f;
        gq[q] = grads + (s >= 0 ? s : 0) * dim;
      }
      for (int t = 0, c = lane; c < dim; c += PA_WAVE, ++t) {
#pragma unroll
        for (int q = 0; q < 4; ++q) acc[q][t] = scq[q] * __half2float(gq[q][c]);
      }
    } else {
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        if (!ok[q]) continue;
        for (int64_t p = lop[q]; p < hip[q]; ++p) {
          const int64_t s = seg_id[perm[p]];
          if (s < 0) continue;
          const float sc = seg_scale ? seg_scale[s] : 1.0f;
          if (sc == 0.0f) continue;
          const __half* g = grads + s * dim;
          for (int t = 0, c = lane; c < dim; c += PA_WAVE, ++t)
            acc[q][t] += __half2float(g[c]) * sc;
        }
      }
    }
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      bool has_nan = false;
      for (int t = 0, c = lane; c < dim; c += PA_WAVE, ++t)
        has_nan |= isnan(acc[q][t]);
      if (ok[q] && __ballot(has_nan) != 0) {
        if (lane == 0) atomicAdd(&skipped[1], 1);
        ok[q] = false;
      }
    }
    if (!ok[0] && !ok[1] && !ok[2] && !ok[3]) continue;
    if (opt == 0) {  // SGD
      for (int t = 0, c = lane; c < dim; c += PA_WAVE, ++t) {
        float w0[4];
#pragma unroll
        for (int q = 0; q < 4; ++q)
          w0[q] = ok[q] ? __builtin_nontemporal_load(&rowk[q][c]) : 0.0f;
#pragma unroll
        for (int q = 0; q < 4; ++q) {
          if (!ok[q]) continue;
          float w2 = w0[q] - p0 * (acc[q][t] + p1 * w0[q]);
          if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
          __builtin_nontemporal_store(w2, &rowk[q][c]);
        }
      }
    } else if (opt == 1) {  // Adagrad
      if (p3 > 0.5f) {  // vectorwise-shared accumulator
        float a0[4], gsq[4] = {};
#pragma unroll
        for (int q = 0; q < 4; ++q) a0[q] = ok[q] ? rowk[q][dim] : 0.0f;
        for (int t = 0, c = lane; c < dim; c += PA_WAVE, ++t) {
          float w0[4];
#pragma unroll
          for (int q = 0; q < 4; ++q)
            w0[q] = ok[q] ? __builtin_nontemporal_load(&rowk[q][c]) : 0.0f;
#pragma unroll
          for (int q = 0; q < 4; ++q) {
            if (!ok[q]) continue;
            float w2 = w0[q] - p0 * acc[q][t] * rsqrtf(a0[q] + p2);
            if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
            __builtin_nontemporal_store(w2, &rowk[q][c]);
            gsq[q] += acc[q][t] * acc[q][t];
          }
        }
#pragma unroll
        for (int off = PA_WAVE / 2; off > 0; off >>= 1)
#pragma unroll
          for (int q = 0; q < 4; ++q) gsq[q] += __shfl_down(gsq[q], off);
        if (lane == 0)
#pragma unroll
          for (int q = 0; q < 4; ++q)
            if (ok[q]) rowk[q][dim] = a0[q] * p1 + gsq[q] / (float)dim;
      } else {
        for (int t = 0, c = lane; c < dim; c += PA_WAVE, ++t) {
          float w0[4], a0[4];
#pragma unroll
          for (int q = 0; q < 4; ++q) {
            a0[q] = ok[q] ? __builtin_nontemporal_load(&rowk[q][dim + c]) : 0.0f;
            w0[q] = ok[q] ? __builtin_nontemporal_load(&rowk[q][c]) : 0.0f;
          }
#pragma unroll
          for (int q = 0; q < 4; ++q) {
            if (!ok[q]) continue;
            float w2 = w0[q] - p0 * acc[q][t] * rsqrtf(a0[q] + p2);
            if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
            __builtin_nontemporal_store(w2, &rowk[q][c]);
            __builtin_nontemporal_store(a0[q] * p1 + acc[q][t] * acc[q][t],
                                        &rowk[q][dim + c]);
          }
        }
      }
    } else {  // Adam
      const float om1 = 1.0f - p1, om2 = 1.0f - p2;
      const float c1 = 1.0f / (1.0f - b1_power), c2 = 1.0f / (1.0f - b2_power);
      for (int t = 0, c = lane; c < dim; c += PA_WAVE, ++t) {
#pragma unroll
        for (int q = 0; q < 4; ++q) {
          if (!ok[q]) continue;
          const float m =
              p1 * __builtin_nontemporal_load(&rowk[q][dim + c]) + om1 * acc[q][t];
          const float v = p2 * __builtin_nontemporal_load(&rowk[q][2 * dim + c]) +
                          om2 * acc[q][t] * acc[q][t];
          const float w0 = __builtin_nontemporal_load(&rowk[q][c]);
          float w2 = w0 - p0 * (m * c1) / (p3 + sqrtf(v * c2));
          if (weight_bound > 0.0f) w2 = fminf(fmaxf(w2, -weight_bound), weight_bound);
          __builtin_nontemporal_store(w2, &rowk[q][c]);
          __builtin_nontemporal_store(m, &rowk[q][dim + c]);
          __builtin_nontemporal_store(v, &rowk[q][2 * dim + c]);
        }
      }
    }
  }
}

inline int n_blocks_for(int64_t work_items, int per_block) {
  int64_t b = (work_items + per_block - 1) / per_block;
  // >> 256 CUs needed to fill the chip; cap and grid-stride beyond
  if (b > 16384) b = 16384;
  if (b < 1) b = 1;
  return (int)b;
}

}  // namespace

// ============================================================ torch bindings

static hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

// u64 radix sort-pairs (keys ascending in UNSIGNED order — exactly the
// dedup/owner-range order; torch.sort would need the sign-flip trick AND
// runs a merge sort with a slow index-iota setup: ~3x the time)
std::vector<torch::Tensor> sort_pairs_u64(torch::Tensor keys) {
  const int64_t n = keys.numel();
  auto opts = torch::TensorOptions().dtype(torch::kInt64).device(keys.device());
  auto iota = torch::empty({n}, opts);
  auto sorted = torch::empty_like(keys);
  auto perm = torch::empty({n}, opts);
  if (n == 0) return {sorted, perm};
  hipStream_t st = cur_stream();
  hipLaunchKernelGGL(iota_i64_kernel, dim3(n_blocks_for(n, 256)), dim3(256),
                     0, st, iota.data_ptr<int64_t>(), n);
  size_t temp_bytes = 0;
  hipcub::DeviceRadixSort::SortPairs(
      nullptr, temp_bytes, (const ull*)keys.data_ptr<int64_t>(),
      (ull*)sorted.data_ptr<int64_t>(),
      (const long long*)iota.data_ptr<int64_t>(),
      (long long*)perm.data_ptr<int64_t>(), n, 0, 64, st);
  auto temp = torch::empty(
      {(int64_t)temp_bytes},
      torch::TensorOptions().dtype(torch::kUInt8).device(keys.device()));
  hipcub::DeviceRadixSort::SortPairs(
      temp.data_ptr(), temp_bytes, (const ull*)keys.data_ptr<int64_t>(),
      (ull*)sorted.data_ptr<int64_t>(),
      (const long long*)iota.data_ptr<int64_t>(),
      (long long*)perm.data_ptr<int64_t>(), n, 0, 64, st);
  return {sorted, perm};
}

torch::Tensor neq_flags(torch::Tensor sorted) {
  const int64_t n = sorted.numel();
  auto flags = torch::empty(
      {n}, torch::TensorOptions().dtype(torch::kBool).device(sorted.device()));
  if (n == 0) return flags;
  hipLaunchKernelGGL(neq_flags_kernel, dim3(n_blocks_for(n, 256)), dim3(256),
                     0, cur_stream(), (const ull*)sorted.data_ptr<int64_t>(),
                     flags.data_ptr<bool>(), n);
  return flags;
}


void store_lookup(torch::Tensor table_keys, torch::Tensor ticks,
                  torch::Tensor arena, torch::Tensor query, torch::Tensor out,
                  int64_t dim, int64_t train, int64_t tick, double lo,
                  double hi, double admit_prob, double state_init,
                  int64_t opt_space, torch::Tensor evict_keys,
                  torch::Tensor evict_count, torch::Tensor evict_rows,
                  torch::Tensor u_count) {
  const int64_t n = query.numel();
  if (n == 0) return;
  const int64_t n_buckets = table_keys.numel() / PA_BUCKET_SIZE;
  const int row_width = (int)(dim + opt_space);
  const bool spill = evict_keys.numel() > 0;
  const long long* n_dev =
      u_count.numel() ? (const long long*)u_count.data_ptr<int64_t>() : nullptr;
  auto opts = torch::TensorOptions().dtype(torch::kInt64).device(query.device());
  auto slots = torch::empty({n}, opts);
  auto is_new = torch::empty({n}, opts.dtype(torch::kInt32));
  torch::Tensor evict_idx;
  if (spill) evict_idx = torch::empty({n}, opts);
  hipStream_t st = cur_stream();
  hipLaunchKernelGGL(probe_claim_kernel, dim3(n_blocks_for(n, 256)), dim3(256),
                     0, st, (ull*)table_keys.data_ptr<int64_t>(),
                     (unsigned*)ticks.data_ptr<int32_t>(),
                     (const ull*)query.data_ptr<int64_t>(), n, n_buckets,
                     (int)train, (unsigned)tick, (float)admit_prob,
                     (long long*)slots.data_ptr<int64_t>(),
                     is_new.data_ptr<int32_t>(),
                     spill ? (ull*)evict_keys.data_ptr<int64_t>() : nullptr,
                     spill ? evict_count.data_ptr<int32_t>() : nullptr,
                     spill ? (long long*)evict_idx.data_ptr<int64_t>() : nullptr,
                     n_dev);
  if (out.scalar_type() == torch::kFloat16) {
    // f16 out: distributed wire path (rows go straight onto the a2a)
    hipLaunchKernelGGL(init_gather_kernel<__half>, dim3(n_blocks_for(n, 4)),
                       dim3(256), 0, st, arena.data_ptr<float>(),
                       (const ull*)query.data_ptr<int64_t>(),
                       (const long long*)slots.data_ptr<int64_t>(),
                       is_new.data_ptr<int32_t>(),
                       (__half*)out.data_ptr<at::Half>(), n, (int)dim,
                       row_width, lo, hi, (float)state_init,
                       spill ? (const long long*)evict_idx.data_ptr<int64_t>() : nullptr,
                       spill ? evict_rows.data_ptr<float>() : nullptr, n_dev);
  } else {
    hipLaunchKernelGGL(init_gather_kernel<float>, dim3(n_blocks_for(n, 4)),
                       dim3(256), 0, st, arena.data_ptr<float>(),
                       (const ull*)query.data_ptr<int64_t>(),
                       (const long long*)slots.data_ptr<int64_t>(),
                       is_new.data_ptr<int32_t>(), out.data_ptr<float>(), n,
                       (int)dim, row_width, lo, hi, (float)state_init,
                       spill ? (const long long*)evict_idx.data_ptr<int64_t>() : nullptr,
                       spill ? evict_rows.data_ptr<float>() : nullptr, n_dev);
  }
}

// fast-path lookup: probe/claim + fused init/f16-scatter to sum rows
// (single-ID groups only — no spill; see init_scatter_kernel)
void store_lookup_sums(torch::Tensor table_keys, torch::Tensor ticks,
                       torch::Tensor arena, torch::Tensor query,
                       torch::Tensor perm, torch::Tensor ustarts,
                       torch::Tensor sums, int64_t dim, int64_t train,
                       int64_t tick, double lo, double hi, double admit_prob,
                       double state_init, int64_t opt_space,
                       torch::Tensor u_count) {
  const int64_t n = query.numel();
  if (n == 0) return;
  const int64_t n_buckets = table_keys.numel() / PA_BUCKET_SIZE;
  const int row_width = (int)(dim + opt_space);
  const long long* n_dev =
      u_count.numel() ? (const long long*)u_count.data_ptr<int64_t>() : nullptr;
  auto opts = torch::TensorOptions().dtype(torch::kInt64).device(query.device());
  auto slots = torch::empty({n}, opts);
  auto is_new = torch::empty({n}, opts.dtype(torch::kInt32));
  hipStream_t st = cur_stream();
  hipLaunchKernelGGL(probe_claim_kernel, dim3(n_blocks_for(n, 256)), dim3(256),
                     0, st, (ull*)table_keys.data_ptr<int64_t>(),
                     (unsigned*)ticks.data_ptr<int32_t>(),
                     (const ull*)query.data_ptr<int64_t>(), n, n_buckets,
                     (int)train, (unsigned)tick, (float)admit_prob,
                     (long long*)slots.data_ptr<int64_t>(),
                     is_new.data_ptr<int32_t>(), nullptr, nullptr, nullptr,
                     n_dev);
  const bool dual = !(dim < PA_WAVE && (PA_WAVE % dim) == 0);
  if (dual) {
    hipLaunchKernelGGL(init_scatter2_kernel,
                       dim3(n_blocks_for((n + 1) / 2, 4)), dim3(256), 0, st,
                       arena.data_ptr<float>(),
                       (const ull*)query.data_ptr<int64_t>(),
                       (const long long*)slots.data_ptr<int64_t>(),
                       is_new.data_ptr<int32_t>(), perm.data_ptr<int64_t>(),
                       ustarts.data_ptr<int64_t>(),
                       (__half*)sums.data_ptr<at::Half>(), n, (int)dim,
                       row_width, lo, hi, (float)state_init, n_dev);
  } else {
    hipLaunchKernelGGL(init_scatter_kernel, dim3(n_blocks_for(n, 4)),
                       dim3(256), 0, st, arena.data_ptr<float>(),
                       (const ull*)query.data_ptr<int64_t>(),
                       (const long long*)slots.data_ptr<int64_t>(),
                       is_new.data_ptr<int32_t>(), perm.data_ptr<int64_t>(),
                       ustarts.data_ptr<int64_t>(),
                       (__half*)sums.data_ptr<at::Half>(), n, (int)dim,
                       row_width, lo, hi, (float)state_init, n_dev);
  }
}

torch::Tensor store_probe(torch::Tensor table_keys, torch::Tensor ticks,
                          torch::Tensor query, int64_t tick,
                          torch::Tensor u_count) {
  const int64_t n = query.numel();
  auto opts = torch::TensorOptions().dtype(torch::kInt64).device(query.device());
  // padded queries: slots defaults to 0 ("found") beyond the device count so
  // the padding never reads as missing
  auto slots = u_count.numel() ? torch::zeros({n}, opts) : torch::empty({n}, opts);
  if (n == 0) return slots;
  auto is_new = torch::empty({n}, opts.dtype(torch::kInt32));
  const int64_t n_buckets = table_keys.numel() / PA_BUCKET_SIZE;
  hipLaunchKernelGGL(probe_claim_kernel, dim3(n_blocks_for(n, 256)), dim3(256),
                     0, cur_stream(), (ull*)table_keys.data_ptr<int64_t>(),
                     (unsigned*)ticks.data_ptr<int32_t>(),
                     (const ull*)query.data_ptr<int64_t>(), n, n_buckets,
                     /*train=*/0, (unsigned)tick, 1.0f,
                     (long long*)slots.data_ptr<int64_t>(),
                     is_new.data_ptr<int32_t>(), nullptr, nullptr, nullptr,
                     u_count.numel()
                         ? (const long long*)u_count.data_ptr<int64_t>()
                         : nullptr);
  return slots;
}

void store_update(torch::Tensor table_keys, torch::Tensor ticks,
                  torch::Tensor arena, torch::Tensor query,
                  torch::Tensor grads, int64_t dim, int64_t opt,
                  std::vector<double> params, double b1_power,
                  double b2_power, double weight_bound,
                  torch::Tensor skipped /* persistent i32[2]: miss, nan */) {
  const int64_t n = query.numel();
  if (n == 0) return;
  const int64_t n_buckets = table_keys.numel() / PA_BUCKET_SIZE;
  const int row_width = (int)arena.size(1);
  hipStream_t st = cur_stream();
  hipLaunchKernelGGL(update_kernel, dim3(n_blocks_for(n, 4)), dim3(256), 0, st,
                     (ull*)table_keys.data_ptr<int64_t>(),
                     (unsigned*)ticks.data_ptr<int32_t>(),
                     arena.data_ptr<float>(),
                     (const ull*)query.data_ptr<int64_t>(),
                     grads.data_ptr<float>(), n, (int)dim, row_width,
                     n_buckets, (int)opt, (float)params[0], (float)params[1],
                     (float)params[2], (float)params[3], (float)b1_power,
                     (float)b2_power, (float)weight_bound,
                     skipped.data_ptr<int32_t>());
}

void store_import(torch::Tensor table_keys, torch::Tensor ticks,
                  torch::Tensor arena, torch::Tensor query, torch::Tensor rows,
                  int64_t tick, torch::Tensor evict_keys,
                  torch::Tensor evict_count, torch::Tensor evict_rows) {
  const int64_t n = query.numel();
  if (n == 0) return;
  const int64_t n_buckets = table_keys.numel() / PA_BUCKET_SIZE;
  const bool spill = evict_keys.numel() > 0;
  hipLaunchKernelGGL(import_kernel, dim3(n_blocks_for(n, 256)), dim3(256), 0,
                     cur_stream(), (ull*)table_keys.data_ptr<int64_t>(),
                     (unsigned*)ticks.data_ptr<int32_t>(),
                     arena.data_ptr<float>(),
                     (const ull*)query.data_ptr<int64_t>(),
                     rows.data_ptr<float>(), n, (int)rows.size(1), n_buckets,
                     (unsigned)tick,
                     spill ? (ull*)evict_keys.data_ptr<int64_t>() : nullptr,
                     spill ? evict_count.data_ptr<int32_t>() : nullptr,
                     spill ? evict_rows.data_ptr<float>() : nullptr);
}

torch::Tensor segment_sum(torch::Tensor rows, torch::Tensor inverse,
                          torch::Tensor seg_offsets, torch::Tensor seg_scale) {
  const int64_t n_seg = seg_offsets.numel() - 1;
  const int64_t dim = rows.size(1);
  auto out = torch::empty(
      {n_seg, dim},
      torch::TensorOptions().dtype(torch::kFloat16).device(rows.device()));
  if (n_seg == 0) return out;
  const float* scale_ptr =
      seg_scale.numel() ? seg_scale.data_ptr<float>() : nullptr;
  hipStream_t st = cur_stream();
  const dim3 grid(n_blocks_for(n_seg, 4)), block(256);
  if (rows.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(segment_sum_kernel<float>, grid, block, 0, st,
                       rows.data_ptr<float>(), inverse.data_ptr<int64_t>(),
                       seg_offsets.data_ptr<int64_t>(), scale_ptr,
                       (__half*)out.data_ptr<at::Half>(), n_seg, (int)dim);
  } else if (rows.scalar_type() == torch::kFloat16) {
    hipLaunchKernelGGL(segment_sum_kernel<__half>, grid, block, 0, st,
                       (const __half*)rows.data_ptr<at::Half>(),
                       inverse.data_ptr<int64_t>(),
                       seg_offsets.data_ptr<int64_t>(), scale_ptr,
                       (__half*)out.data_ptr<at::Half>(), n_seg, (int)dim);
  } else {
    TORCH_CHECK(false, "segment_sum: rows must be f32 or f16");
  }
  return out;
}

void grad_scatter(torch::Tensor grads, torch::Tensor perm,
                  torch::Tensor ustarts, torch::Tensor seg_id,
                  torch::Tensor seg_scale, torch::Tensor out,
                  int64_t accumulate) {
  const int64_t n_unique = ustarts.numel() - 1;
  const int64_t dim = out.size(1);
  if (n_unique == 0) return;
  hipStream_t st = cur_stream();
  const dim3 grid(n_blocks_for(n_unique, 4)), block(256);
  const float* scale_ptr =
      seg_scale.numel() ? seg_scale.data_ptr<float>() : nullptr;
#define PA_GS(T, ACC)                                                         \
  hipLaunchKernelGGL((grad_scatter_kernel<T, ACC>), grid, block, 0, st,       \
                     (const T*)grads.data_ptr(), perm.data_ptr<int64_t>(),    \
                     ustarts.data_ptr<int64_t>(), seg_id.data_ptr<int64_t>(), \
                     scale_ptr, out.data_ptr<float>(),                        \
                     n_unique, (int)dim)
  if (grads.scalar_type() == torch::kFloat16) {
    if (accumulate) PA_GS(__half, true); else PA_GS(__half, false);
  } else if (grads.scalar_type() == torch::kFloat32) {
    if (accumulate) PA_GS(float, true); else PA_GS(float, false);
  } else if (grads.scalar_type() == torch::kBFloat16) {
    if (accumulate) PA_GS(__hip_bfloat16, true); else PA_GS(__hip_bfloat16, false);
  } else {
    TORCH_CHECK(false, "grad_scatter: grads must be f16/bf16/f32");
  }
#undef PA_GS
}

void grad_scatter_idx(torch::Tensor grads, torch::Tensor perm,
                      torch::Tensor ustarts, torch::Tensor seg_id,
                      torch::Tensor seg_scale, torch::Tensor out_idx,
                      torch::Tensor out, torch::Tensor u_count) {
  const int64_t n_unique = out_idx.numel();  // padded upper bound
  const int64_t dim = out.size(1);
  if (n_unique == 0) return;
  TORCH_CHECK(grads.scalar_type() == torch::kFloat16,
              "grad_scatter_idx: f16 grads");
  hipStream_t st = cur_stream();
  const dim3 grid(n_blocks_for(n_unique, 4)), block(256);
  const float* scale_ptr =
      seg_scale.numel() ? seg_scale.data_ptr<float>() : nullptr;
  const long long* n_dev =
      u_count.numel() ? (const long long*)u_count.data_ptr<int64_t>() : nullptr;
  if (out.scalar_type() == torch::kFloat16) {
    hipLaunchKernelGGL(grad_scatter_idx_kernel<__half>, grid, block, 0, st,
                       (const __half*)grads.data_ptr<at::Half>(),
                       perm.data_ptr<int64_t>(), ustarts.data_ptr<int64_t>(),
                       seg_id.data_ptr<int64_t>(), scale_ptr,
                       out_idx.data_ptr<int64_t>(),
                       (__half*)out.data_ptr<at::Half>(), n_unique, (int)dim,
                       n_dev);
  } else if (out.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(grad_scatter_idx_kernel<float>, grid, block, 0, st,
                       (const __half*)grads.data_ptr<at::Half>(),
                       perm.data_ptr<int64_t>(), ustarts.data_ptr<int64_t>(),
                       seg_id.data_ptr<int64_t>(), scale_ptr,
                       out_idx.data_ptr<int64_t>(),
                       out.data_ptr<float>(), n_unique, (int)dim, n_dev);
  } else {
    TORCH_CHECK(false, "grad_scatter_idx: out must be f16/f32");
  }
}

void scatter_update(torch::Tensor table_keys, torch::Tensor ticks,
                    torch::Tensor arena, torch::Tensor uniq,
                    torch::Tensor grads, torch::Tensor perm,
                    torch::Tensor ustarts, torch::Tensor seg_id,
                    torch::Tensor seg_scale, int64_t dim, int64_t opt,
                    std::vector<double> params, double b1_power,
                    double b2_power, double weight_bound,
                    torch::Tensor skipped, torch::Tensor u_count) {
  const int64_t n = uniq.numel();
  if (n == 0) return;
  TORCH_CHECK(grads.scalar_type() == torch::kFloat16, "scatter_update: f16 grads");
  TORCH_CHECK(dim <= 8 * PA_WAVE, "scatter_update: dim too large");
  const int64_t n_buckets = table_keys.numel() / PA_BUCKET_SIZE;
  const int row_width = (int)arena.size(1);
  const float* scale_ptr =
      seg_scale.numel() ? seg_scale.data_ptr<float>() : nullptr;
  // full-wave dims: dual-key variant (2x memory-level parallelism); small
  // dividing dims keep the sub-wave-packed single kernel (more keys/wave)
  const bool packed_small = (dim < PA_WAVE && (PA_WAVE % dim) == 0);
  if (!packed_small && dim <= 128) {
    hipLaunchKernelGGL(scatter_update4_kernel,
                       dim3(n_blocks_for((n + 3) / 4, 4)), dim3(256), 0,
                       cur_stream(), (ull*)table_keys.data_ptr<int64_t>(),
                       (unsigned*)ticks.data_ptr<int32_t>(),
                       arena.data_ptr<float>(),
                       (const ull*)uniq.data_ptr<int64_t>(),
                       (const __half*)grads.data_ptr<at::Half>(),
                       perm.data_ptr<int64_t>(), ustarts.data_ptr<int64_t>(),
                       seg_id.data_ptr<int64_t>(), scale_ptr, n, (int)dim,
                       row_width, n_buckets, (int)opt, (float)params[0],
                       (float)params[1], (float)params[2], (float)params[3],
                       (float)b1_power, (float)b2_power, (float)weight_bound,
                       skipped.data_ptr<int32_t>(),
                       u_count.numel()
                           ? (const long long*)u_count.data_ptr<int64_t>()
                           : nullptr);
    return;
  }
  const bool dual = !packed_small && dim <= 256;
  if (dual) {
    hipLaunchKernelGGL(scatter_update2_kernel,
                       dim3(n_blocks_for((n + 1) / 2, 4)), dim3(256), 0,
                       cur_stream(), (ull*)table_keys.data_ptr<int64_t>(),
                       (unsigned*)ticks.data_ptr<int32_t>(),
                       arena.data_ptr<float>(),
                       (const ull*)uniq.data_ptr<int64_t>(),
                       (const __half*)grads.data_ptr<at::Half>(),
                       perm.data_ptr<int64_t>(), ustarts.data_ptr<int64_t>(),
                       seg_id.data_ptr<int64_t>(), scale_ptr, n, (int)dim,
                       row_width, n_buckets, (int)opt, (float)params[0],
                       (float)params[1], (float)params[2], (float)params[3],
                       (float)b1_power, (float)b2_power, (float)weight_bound,
                       skipped.data_ptr<int32_t>(),
                       u_count.numel()
                           ? (const long long*)u_count.data_ptr<int64_t>()
                           : nullptr);
    return;
  }
  hipLaunchKernelGGL(scatter_update_kernel, dim3(n_blocks_for(n, 4)),
                     dim3(256), 0, cur_stream(),
                     (ull*)table_keys.data_ptr<int64_t>(),
                     (unsigned*)ticks.data_ptr<int32_t>(),
                     arena.data_ptr<float>(),
                     (const ull*)uniq.data_ptr<int64_t>(),
                     (const __half*)grads.data_ptr<at::Half>(),
                     perm.data_ptr<int64_t>(), ustarts.data_ptr<int64_t>(),
                     seg_id.data_ptr<int64_t>(), scale_ptr, n, (int)dim,
                     row_width, n_buckets, (int)opt, (float)params[0],
                     (float)params[1], (float)params[2], (float)params[3],
                     (float)b1_power, (float)b2_power, (float)weight_bound,
                     skipped.data_ptr<int32_t>(),
                     u_count.numel()
                         ? (const long long*)u_count.data_ptr<int64_t>()
                         : nullptr);
}

// -> (send [world*cap+1] zero-padded, idx [n_pad]); overflow accumulates
// into the caller's persistent device counter
std::vector<torch::Tensor> a2a_route(torch::Tensor uniq, torch::Tensor u_count,
                                     int64_t world, int64_t cap,
                                     torch::Tensor overflow) {
  const int64_t n_pad = uniq.numel();
  auto opts = torch::TensorOptions().dtype(torch::kInt64).device(uniq.device());
  auto send = torch::zeros({world * cap + 1}, opts);
  auto idx = torch::empty({n_pad}, opts);
  auto starts = torch::empty({world + 1}, opts);
  hipStream_t st = cur_stream();
  const long long* n_dev =
      u_count.numel() ? (const long long*)u_count.data_ptr<int64_t>() : nullptr;
  hipLaunchKernelGGL(a2a_bounds_kernel, dim3(1), dim3(world + 1), 0, st,
                     (const ull*)uniq.data_ptr<int64_t>(), n_dev, n_pad,
                     (int)world, (long long*)starts.data_ptr<int64_t>());
  if (n_pad)
    hipLaunchKernelGGL(a2a_scatter_kernel, dim3(n_blocks_for(n_pad, 256)),
                       dim3(256), 0, st, (const ull*)uniq.data_ptr<int64_t>(),
                       n_dev, n_pad,
                       (const long long*)starts.data_ptr<int64_t>(), (int)world,
                       cap, (ull*)send.data_ptr<int64_t>(),
                       idx.data_ptr<int64_t>(),
                       (long long*)overflow.data_ptr<int64_t>());
  return {send, idx};
}

torch::Tensor sign_prep(torch::Tensor values, torch::Tensor slot_starts,
                        torch::Tensor prefixes, int64_t spacing) {
  const int64_t n = values.numel();
  auto out = torch::empty_like(values);
  if (n == 0) return out;
  hipLaunchKernelGGL(sign_prep_kernel, dim3(n_blocks_for(n, 256)), dim3(256),
                     0, cur_stream(), (const ull*)values.data_ptr<int64_t>(),
                     slot_starts.data_ptr<int64_t>(),
                     (const ull*)prefixes.data_ptr<int64_t>(),
                     (int)(slot_starts.numel() - 1), (ull)spacing,
                     (ull*)out.data_ptr<int64_t>(), n);
  return out;
}

void dedup_finalize(torch::Tensor svals_flipped, torch::Tensor perm,
                    torch::Tensor neq, torch::Tensor rank, int64_t flip,
                    torch::Tensor inverse, torch::Tensor uniq,
                    torch::Tensor ustarts, torch::Tensor u_count) {
  const int64_t n = svals_flipped.numel();
  if (n == 0) return;
  hipLaunchKernelGGL(dedup_finalize_kernel, dim3(n_blocks_for(n, 256)),
                     dim3(256), 0, cur_stream(),
                     (const ull*)svals_flipped.data_ptr<int64_t>(),
                     perm.data_ptr<int64_t>(), neq.data_ptr<bool>(),
                     rank.data_ptr<int64_t>(), n, (ull)flip,
                     inverse.data_ptr<int64_t>(),
                     (ull*)uniq.data_ptr<int64_t>(),
                     ustarts.data_ptr<int64_t>(),
                     (long long*)u_count.data_ptr<int64_t>());
}

torch::Tensor sign_prep_stack(torch::Tensor values, torch::Tensor in_starts,
                              torch::Tensor out_starts, torch::Tensor prefixes,
                              torch::Tensor rounds, torch::Tensor sizes,
                              torch::Tensor off_starts, torch::Tensor all_offs,
                              int64_t spacing, int64_t n_out) {
  const int64_t n_in = values.numel();
  auto out = torch::empty({n_out}, values.options());
  if (n_in == 0) return out;
  hipLaunchKernelGGL(sign_prep_stack_kernel, dim3(n_blocks_for(n_in, 256)),
                     dim3(256), 0, cur_stream(),
                     (const ull*)values.data_ptr<int64_t>(),
                     in_starts.data_ptr<int64_t>(),
                     out_starts.data_ptr<int64_t>(),
                     (const ull*)prefixes.data_ptr<int64_t>(),
                     rounds.data_ptr<int32_t>(), sizes.data_ptr<int64_t>(),
                     off_starts.data_ptr<int64_t>(),
                     all_offs.data_ptr<int64_t>(),
                     (int)(in_starts.numel() - 1), (ull)spacing,
                     (ull*)out.data_ptr<int64_t>(), n_in);
  return out;
}

void init_dense(pybind11::module_& m);     // csrc/dense.hip
void init_engine(pybind11::module_& m);    // csrc/engine.cpp
void init_interact(pybind11::module_& m);  // csrc/interact.hip

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "persia_amd HIP kernels (gfx950)";
  init_dense(m);
  init_engine(m);
  init_interact(m);
  m.def("store_lookup", &store_lookup, "hash-table lookup/insert + gather");
  m.def("store_probe", &store_probe, "probe-only (spill-tier miss detection)");
  m.def("store_lookup_sums", &store_lookup_sums,
        "fast-path lookup: probe + fused init/f16 scatter to sum rows");
  m.def("store_update", &store_update, "fused sparse optimizer update");
  m.def("store_import", &store_import, "bulk insert rows (checkpoint load)");
  m.def("segment_sum", &segment_sum, "fused gather + per-sample summation");
  m.def("grad_scatter", &grad_scatter, "ordered per-sign gradient scatter");
  m.def("grad_scatter_idx", &grad_scatter_idx,
        "ordered per-sign gradient scatter packed into the a2a send layout");
  m.def("sign_prep", &sign_prep, "prefix-fold + splitmix64 key mixing");
  m.def("sign_prep_stack", &sign_prep_stack,
        "hashstack expansion + prefix-fold + splitmix64 key mixing");
  m.def("dedup_finalize", &dedup_finalize,
        "fixed-shape dedup epilogue (padded uniq/ustarts + device count)");
  m.def("sort_pairs_u64", &sort_pairs_u64, "radix sort-pairs, u64 order");
  m.def("a2a_route", &a2a_route,
        "padded-a2a key routing: range-partition bounds + packed scatter");
  m.def("neq_flags", &neq_flags, "sorted-run boundary flags");
  m.def("scatter_update", &scatter_update,
        "fused ordered grad scatter + optimizer update (no [U,dim] buffer)");
}
