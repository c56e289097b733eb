// Native lookup/update drivers: the per-batch hot path of the embedding
// engine as single C++ calls (the reference's forward/backward engines are
// native Rust; here the orchestration of sign-prep -> dedup -> probe ->
// init/gather -> fused segment-sum (and scatter -> fused optimizer on the
// way back) is C++ driving the HIP kernels + rocPRIM sort via ATen, so the
// pipeline thread does one extension call instead of ~25 Python dispatches).
//
// The distributed path keeps its all_to_all hops in Python (they go through
// torch.distributed process groups); it reuses the pieces exposed here.
#include <torch/extension.h>

// kernels.hip entry points
void store_lookup(torch::Tensor table_keys, torch::Tensor ticks,
                  torch::Tensor arena, torch::Tensor query, torch::Tensor out,
                  int64_t dim, int64_t train, int64_t tick, double lo,
                  double hi, double admit_prob, double state_init,
                  int64_t opt_space, torch::Tensor evict_keys,
                  torch::Tensor evict_count, torch::Tensor evict_rows,
                  torch::Tensor u_count);
void store_lookup_sums(torch::Tensor table_keys, torch::Tensor ticks,
                       torch::Tensor arena, torch::Tensor query,
                       torch::Tensor perm, torch::Tensor ustarts,
                       torch::Tensor sums, int64_t dim, int64_t train,
                       int64_t tick, double lo, double hi, double admit_prob,
                       double state_init, int64_t opt_space,
                       torch::Tensor u_count);
void store_update(torch::Tensor table_keys, torch::Tensor ticks,
                  torch::Tensor arena, torch::Tensor query,
                  torch::Tensor grads, int64_t dim, int64_t opt,
                  std::vector<double> params, double b1_power,
                  double b2_power, double weight_bound, torch::Tensor skipped);
torch::Tensor segment_sum(torch::Tensor rows, torch::Tensor inverse,
                          torch::Tensor seg_offsets, torch::Tensor seg_scale);
void grad_scatter(torch::Tensor grads, torch::Tensor perm,
                  torch::Tensor ustarts, torch::Tensor seg_id,
                  torch::Tensor seg_scale, torch::Tensor out,
                  int64_t accumulate);
torch::Tensor sign_prep(torch::Tensor values, torch::Tensor slot_starts,
                        torch::Tensor prefixes, int64_t spacing);
void scatter_update(torch::Tensor table_keys, torch::Tensor ticks,
                    torch::Tensor arena, torch::Tensor uniq,
                    torch::Tensor grads, torch::Tensor perm,
                    torch::Tensor ustarts, torch::Tensor seg_id,
                    torch::Tensor seg_scale, int64_t dim, int64_t opt,
                    std::vector<double> params, double b1_power,
                    double b2_power, double weight_bound,
                    torch::Tensor skipped, torch::Tensor u_count);
void dedup_finalize(torch::Tensor svals_flipped, torch::Tensor perm,
                    torch::Tensor neq, torch::Tensor rank, int64_t flip,
                    torch::Tensor inverse, torch::Tensor uniq,
                    torch::Tensor ustarts, torch::Tensor u_count);
std::vector<torch::Tensor> sort_pairs_u64(torch::Tensor keys);
torch::Tensor neq_flags(torch::Tensor sorted);

static constexpr int64_t kFlip = std::numeric_limits<int64_t>::min();

// sort-based dedup (torch.sort on GPU = rocPRIM radix/merge sort).
// -> (uniq ascending u64-order, inverse, perm, ustarts)
std::vector<torch::Tensor> dedup_keys(torch::Tensor keys) {
  auto dev = keys.device();
  const int64_t nnz = keys.numel();
  auto opts = torch::TensorOptions().dtype(torch::kInt64).device(dev);
  if (nnz == 0) {
    auto z = torch::zeros({0}, opts);
    return {z, z, z, torch::zeros({1}, opts)};
  }
  auto flipped = keys.bitwise_xor(kFlip);
  auto sorted = flipped.sort();
  auto svals = std::get<0>(sorted);
  auto perm = std::get<1>(sorted);
  auto neq = torch::ones({nnz}, opts.dtype(torch::kBool));
  neq.slice(0, 1, nnz) =
      svals.slice(0, 1, nnz).ne(svals.slice(0, 0, nnz - 1));
  auto inv_sorted = neq.cumsum(0) - 1;
  auto inverse = torch::empty({nnz}, opts);
  inverse.index_put_({perm}, inv_sorted);
  auto uniq = svals.masked_select(neq).bitwise_xor(kFlip);
  auto starts = neq.nonzero().view(-1);
  auto ustarts = torch::cat({starts, torch::full({1}, nnz, opts)});
  return {uniq, inverse, perm, ustarts};
}

// Fixed-shape dedup: every tensor is nnz-padded and the true unique count
// lives ONLY on the device (u_count) — no host synchronization anywhere, so
// the pipeline thread issues the whole batch and returns (a
// masked_select/nonzero dedup forces two stream syncs per batch,
// serializing the producer behind the GPU).  The uniq tail is ZERO-filled
// (the empty-key sentinel, which sign prep never produces), so consumers
// that do scan past u_count probe an impossible key.
// -> {uniq zeros-padded [nnz], inverse [nnz], perm [nnz],
//     ustarts padded [nnz+1], u_count device i64[1]}
std::vector<torch::Tensor> dedup_padded(torch::Tensor keys) {
  const int64_t nnz = keys.numel();
  auto opts = torch::TensorOptions().dtype(torch::kInt64).device(keys.device());
  // u64 radix sort needs no sign-flip: unsigned order IS key/owner order
  auto sp = sort_pairs_u64(keys);
  auto& svals = sp[0];
  auto& perm = sp[1];
  auto neq = neq_flags(svals);
  auto rank = neq.cumsum(0);
  rank.sub_(1);
  auto inverse = torch::empty({nnz}, opts);
  auto uniq = torch::zeros({nnz}, opts);
  auto ustarts = torch::empty({nnz + 1}, opts);
  auto u_count = torch::empty({1}, opts);
  dedup_finalize(svals, perm, neq, rank, /*flip=*/0, inverse, uniq, ustarts,
                 u_count);
  return {uniq, inverse, perm, ustarts, u_count};
}

// Single-GPU fused lookup: raw values -> per-sample summed embeddings.
// Returns {sums f16 [n_segs, dim], uniq_keys, inverse, perm, ustarts}.
std::vector<torch::Tensor> lookup_local(
    torch::Tensor values, torch::Tensor slot_starts, torch::Tensor prefixes,
    int64_t spacing, torch::Tensor cat_offsets, torch::Tensor seg_scale,
    torch::Tensor table_keys, torch::Tensor ticks, torch::Tensor arena,
    int64_t dim, int64_t train, int64_t tick, double lo, double hi,
    double admit_prob, double state_init, int64_t opt_space) {
  auto keys = sign_prep(values, slot_starts, prefixes, spacing);
  auto d = dedup_padded(keys);
  auto& uniq = d[0];
  auto& inverse = d[1];
  auto& perm = d[2];
  auto& ustarts = d[3];
  auto& u_count = d[4];
  const int64_t nnz = keys.numel();
  auto dev = values.device();
  auto opts = torch::TensorOptions().dtype(torch::kInt64).device(dev);
  // the all-single-ID fast path has exactly one id per segment
  // (cat_offsets is the identity partition), so the sum rows are a pure
  // per-position scatter of the unique rows: fuse init+gather+f16-cast into
  // one kernel and skip the [nnz, dim] f32 intermediate entirely
  auto sums = torch::empty(
      {nnz, dim}, torch::TensorOptions().dtype(torch::kFloat16).device(dev));
  store_lookup_sums(table_keys, ticks, arena, uniq, perm, ustarts, sums, dim,
                    train, tick, lo, hi, admit_prob, state_init, opt_space,
                    u_count);
  return {sums, uniq, inverse, perm, ustarts, u_count};
}

// Single-GPU fused backward: per-segment grads -> per-sign scatter -> fused
// optimizer update on the shard.
void update_local(torch::Tensor grads, torch::Tensor perm,
                  torch::Tensor ustarts, torch::Tensor seg_id,
                  torch::Tensor seg_scale, torch::Tensor uniq_keys,
                  torch::Tensor table_keys, torch::Tensor ticks,
                  torch::Tensor arena, int64_t dim, int64_t opt,
                  std::vector<double> params, double b1_power, double b2_power,
                  double weight_bound, torch::Tensor skipped,
                  torch::Tensor u_count) {
  if (grads.scalar_type() == torch::kFloat16 && dim <= 512) {
    // single fused kernel: ordered scatter + optimizer, no [U,dim] buffer
    scatter_update(table_keys, ticks, arena, uniq_keys, grads, perm, ustarts,
                   seg_id, seg_scale, dim, opt, params, b1_power, b2_power,
                   weight_bound, skipped, u_count);
    return;
  }
  // rare fallback (f32 grads or dim > 512): needs exact shapes — for a
  // padded (device-count) dedup, pay the one host sync and slice
  if (u_count.numel()) {
    const int64_t U = u_count.item<int64_t>();
    uniq_keys = uniq_keys.narrow(0, 0, U);
    ustarts = ustarts.narrow(0, 0, U + 1);
  }
  auto buf = torch::empty(
      {uniq_keys.numel(), dim},
      torch::TensorOptions().dtype(torch::kFloat32).device(grads.device()));
  grad_scatter(grads, perm, ustarts, seg_id, seg_scale, buf, /*accumulate=*/0);
  store_update(table_keys, ticks, arena, uniq_keys, buf, dim, opt, params,
               b1_power, b2_power, weight_bound, skipped);
}


// ---------------------------------------------------------------- host tier
// Host-DRAM spill tier (native): rows evicted from the HBM table park here
// and come back on their next lookup.  Exclusive bounded LRU with
// insert-order recency, matching persia_amd/core/store.py HostTier (the
// python oracle) exactly: re-insert refreshes recency, fetch removes, the
// oldest entry is dropped past capacity.  Replaces per-key python dict
// loops in the pipeline thread (reference: remote CPU parameter-server
// shards, embedding_parameter_server/lib.rs).
#include <list>
#include <memory>
#include <unordered_map>
#include <vector>

struct NativeHostTier {
  // Row payloads live in fixed-size slabs with a free-list (two heap
  // allocations PER INSERT — list node + row vector — dominated the
  // original implementation and fragmented the host heap at 1e8-row
  // scale); the key map is sharded 16 ways so rehash pauses are 1/16th
  // the size.  LRU stays one global list of (key, slot).
  static constexpr int64_t CHUNK_ROWS = 1 << 16;
  static constexpr int SHARDS = 16;
  int64_t capacity, row_width;
  std::vector<std::unique_ptr<float[]>> chunks;
  std::vector<int64_t> free_slots;
  int64_t next_fresh = 0;  // slots handed out so far (before any free)
  std::list<std::pair<uint64_t, int64_t>> lru;  // (key, slot); front = oldest
  std::unordered_map<uint64_t,
                     std::list<std::pair<uint64_t, int64_t>>::iterator>
      map[SHARDS];

  NativeHostTier(int64_t cap, int64_t rw) : capacity(cap), row_width(rw) {}

  float* rowp(int64_t slot) {
    return chunks[slot / CHUNK_ROWS].get() + (slot % CHUNK_ROWS) * row_width;
  }
  int64_t alloc_slot() {
    if (!free_slots.empty()) {
      const int64_t s = free_slots.back();
      free_slots.pop_back();
      return s;
    }
    if (next_fresh >= (int64_t)chunks.size() * CHUNK_ROWS)
      chunks.emplace_back(new float[CHUNK_ROWS * row_width]);
    return next_fresh++;
  }
  auto& shard(uint64_t key) { return map[key & (SHARDS - 1)]; }

  void erase_entry(std::list<std::pair<uint64_t, int64_t>>::iterator it) {
    free_slots.push_back(it->second);
    shard(it->first).erase(it->first);
    lru.erase(it);
  }

  void insert(torch::Tensor keys, torch::Tensor rows) {
    TORCH_CHECK(!keys.is_cuda() && !rows.is_cuda(), "HostTier: cpu tensors");
    TORCH_CHECK(rows.size(1) == row_width, "HostTier: row width mismatch");
    auto kc = keys.contiguous();
    auto rc = rows.contiguous();
    const int64_t* kp = kc.data_ptr<int64_t>();
    const float* rp = rc.data_ptr<float>();
    const int64_t k = kc.numel();
    for (int64_t i = 0; i < k; ++i) {
      const uint64_t key = (uint64_t)kp[i];
      auto& m = shard(key);
      auto it = m.find(key);
      int64_t slot;
      if (it != m.end()) {  // refresh: reuse the slot, move to back
        slot = it->second->second;
        lru.erase(it->second);
        lru.emplace_back(key, slot);
        it->second = std::prev(lru.end());
      } else {
        slot = alloc_slot();
        lru.emplace_back(key, slot);
        m[key] = std::prev(lru.end());
      }
      std::copy(rp + i * row_width, rp + (i + 1) * row_width, rowp(slot));
    }
    while ((int64_t)lru.size() > capacity) erase_entry(lru.begin());
  }

  std::tuple<torch::Tensor, torch::Tensor> fetch(torch::Tensor keys) {
    auto kc = keys.contiguous();
    const int64_t* kp = kc.data_ptr<int64_t>();
    const int64_t k = kc.numel();
    auto rows = torch::zeros({k, row_width}, torch::kFloat32);
    auto found = torch::zeros({k}, torch::kBool);
    float* rp = rows.data_ptr<float>();
    bool* fp = found.data_ptr<bool>();
    for (int64_t i = 0; i < k; ++i) {
      const uint64_t key = (uint64_t)kp[i];
      auto& m = shard(key);
      auto it = m.find(key);
      if (it == m.end()) continue;
      const float* src = rowp(it->second->second);
      std::copy(src, src + row_width, rp + i * row_width);
      fp[i] = true;
      erase_entry(it->second);
    }
    return {rows, found};
  }

  std::tuple<torch::Tensor, torch::Tensor> export_all() {
    const int64_t n = (int64_t)lru.size();
    auto keys = torch::empty({n}, torch::kInt64);
    auto rows = torch::empty({n, row_width}, torch::kFloat32);
    int64_t* kp = keys.data_ptr<int64_t>();
    float* rp = rows.data_ptr<float>();
    int64_t i = 0;
    for (auto& e : lru) {
      kp[i] = (int64_t)e.first;
      const float* src = rowp(e.second);
      std::copy(src, src + row_width, rp + i * row_width);
      ++i;
    }
    return {keys, rows};
  }

  int64_t size() const { return (int64_t)lru.size(); }
  bool contains(int64_t key) {
    return shard((uint64_t)key).count((uint64_t)key) != 0;
  }
  void clear() {
    lru.clear();
    for (auto& m : map) m.clear();
    chunks.clear();
    free_slots.clear();
    next_fresh = 0;
  }
};

void init_engine(pybind11::module_& m) {
  m.def("dedup_keys", &dedup_keys, "sort-based dedup of u64 keys");
  m.def("dedup_padded", &dedup_padded,
        "sync-free dedup: nnz-padded outputs + device unique count");
  pybind11::class_<NativeHostTier>(m, "HostTier")
      .def(pybind11::init<int64_t, int64_t>())
      .def("insert", &NativeHostTier::insert)
      .def("fetch", &NativeHostTier::fetch)
      .def("export_all", &NativeHostTier::export_all)
      .def("size", &NativeHostTier::size)
      .def("contains", &NativeHostTier::contains)
      .def("clear", &NativeHostTier::clear);
  m.def("lookup_local", &lookup_local,
        "fused single-GPU lookup (sign prep + dedup + probe + gather + sum)");
  m.def("update_local", &update_local,
        "fused single-GPU backward (scatter + optimizer update)");
}
