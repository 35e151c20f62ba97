// ev_kernels.hip — MI355X (gfx950, CDNA4) embedding-engine kernels.
//
// The HBM embedding engine: an open-addressing (linear probe) hash table
// keyed by int64 feature id mapping to a monotonically-allocated int32
// value-slot, with per-entry frequency/version metadata for feature
// admission (counter filter) and eviction. Value rows live in a dense
// [max_slots, dim] fp32 slab; optimizer states are parallel slabs indexed
// by the same slots.
//
// Capability parity (not a port) with the reference engine:
//   - hash lookup/insert  ≙ GPUHashTable/cuco dynamic_map probing
//     (reference: gpu_hash_table.cu.cc:260-541) — re-designed as a single
//     flat power-of-two table with 64-wide-wavefront-friendly per-thread
//     probes; keys are pre-uniqued on the host side of the step so probes
//     are contention-free except first-insert CAS.
//   - fused gather+segment-pool ≙ EmbeddingLookUp + ApplyCombiner
//     (reference: fused_embedding_local_ops_gpu.cu.cc:42, SumUpEmbeddingShard)
//   - grad scatter ≙ DoEmbeddingGrad/DistributeGradToShard
//   - sparse optimizer applies ≙ training_ali_ops_gpu.cu.cc kernels,
//     operating directly on slot indices from the step's single probe
//     (the reference's `_OPT_` indices-as-pointers fusion).
//
// CDNA4 notes: wavefront = 64; all elementwise kernels use a (row, dim)
// thread mapping with grid-stride so consecutive lanes read consecutive
// value-row elements (coalesced); blocks are multiples of 64; atomics are
// device-scope by default (per-XCD L2 non-coherence is safe).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

// current-stream accessor (ROCm ATen name)
static inline hipStream_t current_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

#include <cstdint>

#define DEV_INLINE __device__ __forceinline__

static constexpr int64_t EMPTY_KEY = INT64_MIN;  // reserved sentinel
static constexpr int64_t PAD_KEY = INT64_MAX;    // wire-padding sentinel
static constexpr int kBlock = 256;

namespace {

DEV_INLINE uint64_t mix_hash(uint64_t k) {
  // splitmix64 finalizer — good avalanche for sequential ids
  k += 0x9E3779B97F4A7C15ull;
  k = (k ^ (k >> 30)) * 0xBF58476D1CE4E5B9ull;
  k = (k ^ (k >> 27)) * 0x94D049BB133111EBull;
  return k ^ (k >> 31);
}

DEV_INLINE float bf2f(__hip_bfloat16 v) { return __bfloat162float(v); }
DEV_INLINE __hip_bfloat16 f2bf(float v) { return __float2bfloat16(v); }

// ---------------------------------------------------------------------
// hash probe / insert
// ---------------------------------------------------------------------

// Lookup-or-insert for a batch of UNIQUE keys. Per key:
//   - find or claim (CAS) a hash entry
//   - freq += count; version = step (no atomics needed: keys unique)
//   - admit a value slot once freq >= filter_freq; initialize the value
//     row from default_values[key % default_value_dim]
// out_slots[i] = slot or -1 (not admitted).
// NOTE every kernel whose grid n_blocks() caps at 65535 blocks MUST
// grid-stride: a plain `if (i >= n) return` silently dropped the tail of
// batches beyond 16.7M keys (found when a 120M-key rebalance rebuild
// lost most of the table).
__global__ void k_lookup_insert(
    const int64_t* __restrict__ keys, const int32_t* __restrict__ counts,
    int n, int64_t* __restrict__ ht_keys, int32_t* __restrict__ ht_slot,
    int32_t* __restrict__ ht_freq, int64_t* __restrict__ ht_version,
    int64_t cap_mask, int32_t* __restrict__ slot_counter,
    int32_t* __restrict__ entry_counter, int max_slots,
    float* __restrict__ values, const float* __restrict__ default_values,
    int dim, int default_value_dim, int key_bits, int init_limit,
    int filter_freq, int64_t step, int train,
    int32_t* __restrict__ out_slots, int32_t* __restrict__ error_flag) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t gstride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += gstride) {
  const int64_t key = keys[i];
  uint64_t h = mix_hash((uint64_t)key) & (uint64_t)cap_mask;
  int32_t slot = -1;
  for (int64_t probe = 0; probe <= cap_mask; ++probe) {
    int64_t idx = (int64_t)((h + probe) & (uint64_t)cap_mask);
    int64_t cur = ht_keys[idx];
    if (cur != key) {
      if (cur != EMPTY_KEY) continue;
      if (!train) { out_slots[i] = -1; goto done_i; }
      int64_t prev = (int64_t)atomicCAS(
          (unsigned long long*)&ht_keys[idx],
          (unsigned long long)EMPTY_KEY, (unsigned long long)key);
      if (prev != EMPTY_KEY && prev != key) continue;  // lost race
      if (prev == EMPTY_KEY) atomicAdd(entry_counter, 1);
    }
    // found or claimed entry idx
    int32_t freq = ht_freq[idx];
    if (train) {
      freq += (counts ? counts[i] : 1);
      ht_freq[idx] = freq;
      ht_version[idx] = step;
    }
    slot = ht_slot[idx];
    if (slot < 0 && train && freq >= filter_freq) {
      slot = atomicAdd(slot_counter, 1);
      if (slot >= max_slots) {  // out of slab space: host must grow
        atomicExch(error_flag, 1);
        out_slots[i] = -1;
        goto done_i;
      }
      ht_slot[idx] = slot;
      // multi-tier (HBM_DRAM): rows at slot >= init_limit live in the host
      // cold slab; the host initializes those (kernel must not touch them)
      if (slot >= init_limit) {
        out_slots[i] = slot;
        goto done_i;
      }
      // composite keys (EmbeddingCollection): default row is
      // table * dvd + (raw_key % dvd); key_bits == 0 means plain keys
      int64_t dvrow;
      if (key_bits > 0) {
        int64_t mask = ((int64_t)1 << key_bits) - 1;
        dvrow = (key >> key_bits) * default_value_dim +
                (int64_t)((uint64_t)(key & mask) %
                          (uint64_t)default_value_dim);
      } else {
        dvrow = (int64_t)((uint64_t)key % (uint64_t)default_value_dim);
      }
      const float* src = default_values + dvrow * dim;
      float* dst = values + (int64_t)slot * dim;
      for (int d = 0; d < dim; ++d) dst[d] = src[d];
    }
    out_slots[i] = slot;
    goto done_i;
  }
  atomicExch(error_flag, 2);  // table full (host sizing bug)
  out_slots[i] = -1;
  done_i:;
  }
}

// Bulk import used by restore: entries are created admitted with the given
// slot ids (slots pre-assigned densely by the host).
__global__ void k_insert_bulk(
    const int64_t* __restrict__ keys, const int32_t* __restrict__ slots,
    const int32_t* __restrict__ freqs, const int64_t* __restrict__ versions,
    int n, int64_t* __restrict__ ht_keys, int32_t* __restrict__ ht_slot,
    int32_t* __restrict__ ht_freq, int64_t* __restrict__ ht_version,
    int64_t cap_mask, int32_t* __restrict__ entry_counter,
    int32_t* __restrict__ error_flag) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t gstride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += gstride) {
    const int64_t key = keys[i];
    uint64_t h = mix_hash((uint64_t)key) & (uint64_t)cap_mask;
    bool placed = false;
    for (int64_t probe = 0; probe <= cap_mask && !placed; ++probe) {
      int64_t idx = (int64_t)((h + probe) & (uint64_t)cap_mask);
      int64_t cur = ht_keys[idx];
      if (cur != key) {
        if (cur != EMPTY_KEY) continue;
        int64_t prev = (int64_t)atomicCAS(
            (unsigned long long*)&ht_keys[idx],
            (unsigned long long)EMPTY_KEY, (unsigned long long)key);
        if (prev != EMPTY_KEY && prev != key) continue;
        if (prev == EMPTY_KEY) atomicAdd(entry_counter, 1);
      }
      ht_slot[idx] = slots[i];
      if (freqs) ht_freq[idx] = freqs[i];
      if (versions) ht_version[idx] = versions[i];
      placed = true;
    }
    if (!placed) atomicExch(error_flag, 2);
  }
}

// ---------------------------------------------------------------------
// fused hash dedup — replaces sort-based unique on the training hot path
// ---------------------------------------------------------------------
// The hash table itself deduplicates: pass A runs once per OCCURRENCE,
// inserting the key if absent, counting frequency exactly, and claiming a
// compact index [0, m) for the first toucher of this epoch (atomicExch on
// a per-entry epoch stamp — exactly one winner). Pass B (per unique key)
// applies admission + default-value init. Pass C (per occurrence) reads
// back inverse indices and per-batch counts. Equivalent outputs to
// torch.unique(return_inverse, return_counts) + the probe, without any
// sort (the rocprim merge sorts were ~15% of the DLRM step).

__global__ void k_dedup_pass_a(
    const int64_t* __restrict__ keys, int nnz, int64_t* __restrict__ ht_keys,
    int32_t* __restrict__ ht_freq, int64_t* __restrict__ ht_version,
    int32_t* __restrict__ ht_epoch, int32_t* __restrict__ ht_compact,
    int64_t cap_mask, int epoch, int64_t step,
    int32_t* __restrict__ entry_counter, int32_t* __restrict__ m_counter,
    int64_t* __restrict__ uniq_keys, int64_t* __restrict__ compact_entry,
    int64_t* __restrict__ occ_entry, int32_t* __restrict__ error_flag) {
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; j < nnz; j += stride) {
    const int64_t key = keys[j];
    uint64_t h = mix_hash((uint64_t)key) & (uint64_t)cap_mask;
    for (int64_t probe = 0; probe <= cap_mask; ++probe) {
      int64_t idx = (int64_t)((h + probe) & (uint64_t)cap_mask);
      int64_t cur = ht_keys[idx];
      if (cur != key) {
        if (cur != EMPTY_KEY) continue;
        int64_t prev = (int64_t)atomicCAS(
            (unsigned long long*)&ht_keys[idx],
            (unsigned long long)EMPTY_KEY, (unsigned long long)key);
        if (prev != EMPTY_KEY && prev != key) continue;
        if (prev == EMPTY_KEY) atomicAdd(entry_counter, 1);
      }
      // freq/version updates moved to pass B (one per UNIQUE key via
      // pass C's exact counts): hot keys were serializing thousands of
      // per-occurrence atomics on one cache line here. The plain read
      // short-circuits the epoch claim for already-claimed keys; racing
      // first-touchers are resolved by the atomicExch.
      if (ht_epoch[idx] != epoch) {
        int old = atomicExch(&ht_epoch[idx], epoch);
        if (old != epoch) {  // first toucher this step
          int c = atomicAdd(m_counter, 1);
          ht_compact[idx] = c;
          uniq_keys[c] = key;
          compact_entry[c] = idx;
        }
      }
      occ_entry[j] = idx;  // pass C reads compact ids WITHOUT re-probing
      goto next_j;
    }
    atomicExch(error_flag, 2);
  next_j:;
  }
}

// Graph-capture pass A: epoch and step advance on-device (bumped by
// k_bump_epoch inside the captured step) so replays keep deduplicating
// correctly with zero host involvement.
__global__ void k_bump_epoch(int32_t* __restrict__ epoch_dev,
                             int64_t* __restrict__ step_dev) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    epoch_dev[0] += 1;
    step_dev[0] += 1;
  }
}

__global__ void k_dedup_pass_a_dev(
    const int64_t* __restrict__ keys, int nnz, int64_t* __restrict__ ht_keys,
    int32_t* __restrict__ ht_freq, int64_t* __restrict__ ht_version,
    int32_t* __restrict__ ht_epoch, int32_t* __restrict__ ht_compact,
    int64_t cap_mask, const int32_t* __restrict__ epoch_dev,
    const int64_t* __restrict__ step_dev,
    int32_t* __restrict__ entry_counter, int32_t* __restrict__ m_counter,
    int64_t* __restrict__ uniq_keys, int64_t* __restrict__ compact_entry,
    int64_t* __restrict__ occ_entry, int32_t* __restrict__ error_flag) {
  const int epoch = *epoch_dev;
  const int64_t step = *step_dev;
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; j < nnz; j += stride) {
    const int64_t key = keys[j];
    uint64_t h = mix_hash((uint64_t)key) & (uint64_t)cap_mask;
    for (int64_t probe = 0; probe <= cap_mask; ++probe) {
      int64_t idx = (int64_t)((h + probe) & (uint64_t)cap_mask);
      int64_t cur = ht_keys[idx];
      if (cur != key) {
        if (cur != EMPTY_KEY) continue;
        int64_t prev = (int64_t)atomicCAS(
            (unsigned long long*)&ht_keys[idx],
            (unsigned long long)EMPTY_KEY, (unsigned long long)key);
        if (prev != EMPTY_KEY && prev != key) continue;
        if (prev == EMPTY_KEY) atomicAdd(entry_counter, 1);
      }
      // freq/version handled per-unique in pass B (see k_dedup_pass_a)
      if (ht_epoch[idx] != epoch) {
        int old = atomicExch(&ht_epoch[idx], epoch);
        if (old != epoch) {
          int c = atomicAdd(m_counter, 1);
          ht_compact[idx] = c;
          uniq_keys[c] = key;
          compact_entry[c] = idx;
        }
      }
      occ_entry[j] = idx;
      goto next_j2;
    }
    atomicExch(error_flag, 2);
  next_j2:;
  }
}

// Pass B (per unique key): freq/version bookkeeping (one write per
// unique key, using pass C's exact per-batch counts — runs AFTER pass
// C), then admission + default-value init, slots out.
__global__ void k_dedup_pass_b(
    const int64_t* __restrict__ compact_entry,
    const int64_t* __restrict__ uniq_keys, int m,
    int32_t* __restrict__ ht_slot, int32_t* __restrict__ ht_freq,
    int64_t* __restrict__ ht_version,
    const int32_t* __restrict__ batch_counts, int64_t step,
    int32_t* __restrict__ slot_counter, int max_slots,
    float* __restrict__ values, const float* __restrict__ default_values,
    int dim, int default_value_dim, int key_bits, int init_limit,
    int filter_freq, int32_t* __restrict__ out_slots,
    int32_t* __restrict__ error_flag) {
  int64_t c = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t gstride = gridDim.x * (int64_t)blockDim.x;
  for (; c < m; c += gstride) {
  int64_t idx = compact_entry[c];
  ht_freq[idx] += batch_counts[c];
  ht_version[idx] = step;
  int32_t slot = ht_slot[idx];
  if (slot < 0 && ht_freq[idx] >= filter_freq) {
    slot = atomicAdd(slot_counter, 1);
    if (slot >= max_slots) {
      atomicExch(error_flag, 1);
      out_slots[c] = -1;
      continue;
    }
    ht_slot[idx] = slot;
    if (slot < init_limit) {
      const int64_t key = uniq_keys[c];
      int64_t dvrow;
      if (key_bits > 0) {
        int64_t mask = ((int64_t)1 << key_bits) - 1;
        dvrow = (key >> key_bits) * default_value_dim +
                (int64_t)((uint64_t)(key & mask) %
                          (uint64_t)default_value_dim);
      } else {
        dvrow = (int64_t)((uint64_t)key % (uint64_t)default_value_dim);
      }
      const float* src = default_values + dvrow * dim;
      float* dst = values + (int64_t)slot * dim;
      for (int d = 0; d < dim; ++d) dst[d] = src[d];
    }
  }
  out_slots[c] = slot;
  }
}

// Padded pass B for hipGraph capture: grid covers n_cap (= nnz upper
// bound); the true unique count comes from the device counter so no host
// sync happens inside the captured step. Tail entries get slot -1.
__global__ void k_dedup_pass_b_padded(
    const int64_t* __restrict__ compact_entry,
    const int64_t* __restrict__ uniq_keys, int n_cap,
    const int32_t* __restrict__ m_dev, int32_t* __restrict__ ht_slot,
    int32_t* __restrict__ ht_freq, int64_t* __restrict__ ht_version,
    const int32_t* __restrict__ batch_counts,
    const int64_t* __restrict__ step_dev,
    int32_t* __restrict__ slot_counter,
    int max_slots, float* __restrict__ values,
    const float* __restrict__ default_values, int dim,
    int default_value_dim, int key_bits, int init_limit, int filter_freq,
    int32_t* __restrict__ out_slots, int32_t* __restrict__ error_flag) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= n_cap) return;
  if (c >= *m_dev || uniq_keys[c] == PAD_KEY) {
    // tail slot, or the wire-padding sentinel of the sharded exchange:
    // no admission, no metadata (PAD_KEY's composite-decomposed default
    // row would read far out of bounds of default_values)
    out_slots[c] = -1;
    return;
  }
  int64_t idx = compact_entry[c];
  ht_freq[idx] += batch_counts[c];
  ht_version[idx] = *step_dev;
  int32_t slot = ht_slot[idx];
  if (slot < 0 && ht_freq[idx] >= filter_freq) {
    slot = atomicAdd(slot_counter, 1);
    if (slot >= max_slots) {
      atomicExch(error_flag, 1);
      out_slots[c] = -1;
      return;
    }
    ht_slot[idx] = slot;
    if (slot < init_limit) {
      const int64_t key = uniq_keys[c];
      int64_t dvrow;
      if (key_bits > 0) {
        int64_t mask = ((int64_t)1 << key_bits) - 1;
        dvrow = (key >> key_bits) * default_value_dim +
                (int64_t)((uint64_t)(key & mask) %
                          (uint64_t)default_value_dim);
      } else {
        dvrow = (int64_t)((uint64_t)key % (uint64_t)default_value_dim);
      }
      const float* src = default_values + dvrow * dim;
      float* dst = values + (int64_t)slot * dim;
      for (int d = 0; d < dim; ++d) dst[d] = src[d];
    }
  }
  out_slots[c] = slot;
}

// Zero-fill (float4-wide). Used instead of hipMemsetAsync because memset
// nodes recorded during hipGraph capture were observed NOT to replay
// (counts/grad buffers accumulated across replays -> OOB scatters).
__global__ void k_zero_f32(float* __restrict__ p, int64_t n) {
  int64_t i = (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) * 4;
  int64_t stride = gridDim.x * (int64_t)blockDim.x * 4;
  for (; i + 3 < n; i += stride)
    *reinterpret_cast<float4*>(p + i) = make_float4(0.f, 0.f, 0.f, 0.f);
  if (blockIdx.x == 0 && threadIdx.x == 0)
    for (int64_t j = n & ~3LL; j < n; ++j) p[j] = 0.0f;
}

// ---------------------------------------------------------------------
// padded peer routing — the captured distributed exchange
// ---------------------------------------------------------------------
// The sharded collection's all-to-all must have static shapes to live
// inside a hipGraph (RCCL collectives capture, but only with fixed
// splits). Every peer slot is padded to `cap` rows; pad keys are
// PAD_KEY, which the owner deduplicates into one (ignored) entry, so no
// count exchange is needed on the wire at all (≙ the reference's
// two-phase count+payload protocol, all2all_input_dispatcher.cu:250-280,
// re-designed shape-static for graph replay).

// Bump only the dedup epoch (NOT the step): the distributed step runs
// TWO dedups per training step against the same table (requester-side
// claim-only dedup, then owner-side dedup of the padded exchange), and
// version/eviction semantics must advance once per step.
__global__ void k_bump_epoch_only(int32_t* __restrict__ epoch_dev) {
  if (threadIdx.x == 0 && blockIdx.x == 0) epoch_dev[0] += 1;
}

// Reset the padded send layout: keys = PAD_KEY, counts = 0, cursors = 0.
__global__ void k_route_fill(int64_t* __restrict__ send_keys,
                             int32_t* __restrict__ send_cnt,
                             int32_t* __restrict__ peer_cursor, int total,
                             int world) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < total; i += stride) {
    send_keys[i] = PAD_KEY;
    send_cnt[i] = 0;
    if (i < world) peer_cursor[i] = 0;
  }
}

// Scatter the step's unique keys into per-peer padded blocks.
// owner = (key & key_mask) % world (the raw id routes; composite table
// tags are stripped). route_pos[u] remembers where unique u landed so
// the returned embedding rows / outgoing grad rows can be addressed
// without any host-side ordering. Overflowing a peer's cap sets
// error_flag = 4 (host must re-capture with a larger cap).
//
// Cursor reservation is BLOCK-AGGREGATED: per-owner counts collect in
// LDS (fast bank atomics), then one global atomicAdd per (block, owner)
// reserves a range. A naive per-element global atomic serialized on the
// per-owner cache line (measured 525 us at world=1; this form is ~30 us).
static constexpr int kMaxWorld = 64;

__global__ void k_route_pad(
    const int64_t* __restrict__ uniq_keys, const int32_t* __restrict__ counts,
    const int32_t* __restrict__ m_dev, int n_cap, int world, int cap,
    int key_bits, int64_t* __restrict__ send_keys,
    int32_t* __restrict__ send_cnt, int32_t* __restrict__ route_pos,
    int32_t* __restrict__ peer_cursor, int32_t* __restrict__ error_flag) {
  const int m = min(*m_dev, n_cap);
  const int64_t key_mask =
      key_bits > 0 ? (((int64_t)1 << key_bits) - 1) : ~(int64_t)0;
  __shared__ int32_t lcnt[kMaxWorld];
  __shared__ int32_t lbase[kMaxWorld];
  // grid-stride by whole blocks so every thread of a block participates
  // in the aggregation barriers
  for (int64_t base = (int64_t)blockIdx.x * blockDim.x; base < m;
       base += (int64_t)gridDim.x * blockDim.x) {
    int64_t u = base + threadIdx.x;
    for (int o = threadIdx.x; o < world; o += blockDim.x) lcnt[o] = 0;
    __syncthreads();
    int owner = -1, my = 0;
    int64_t key = 0;
    if (u < m) {
      key = uniq_keys[u];
      owner = (int)((uint64_t)(key & key_mask) % (uint64_t)world);
      my = atomicAdd(&lcnt[owner], 1);
    }
    __syncthreads();
    for (int o = threadIdx.x; o < world; o += blockDim.x)
      lbase[o] = lcnt[o] ? atomicAdd(&peer_cursor[o], lcnt[o]) : 0;
    __syncthreads();
    if (u < m) {
      int pos = lbase[owner] + my;
      if (pos >= cap) {
        atomicExch(error_flag, 4);
        route_pos[u] = owner * cap;  // in-bounds dummy; run is poisoned
      } else {
        int64_t o = (int64_t)owner * cap + pos;
        send_keys[o] = key;
        send_cnt[o] = counts[u];
        route_pos[u] = (int32_t)o;
      }
    }
    __syncthreads();
  }
}

// Owner-side helpers, PAD-aware. The pads are the scaling hazard: every
// pad element maps to ONE unique (PAD_KEY), so generic index_add /
// counting kernels serialize hundreds of thousands of atomics on a
// single cache line (measured: torch index_add 783 us/step, pass C
// 161 us). All kernels below skip pad elements outright.

// Pass C variant that marks pads: inverse[j] = -1, no counts atomic.
__global__ void k_dedup_pass_c_pad(
    const int64_t* __restrict__ keys, int nnz,
    const int64_t* __restrict__ ht_keys,
    const int32_t* __restrict__ ht_compact, int64_t cap_mask,
    int32_t* __restrict__ inverse, int32_t* __restrict__ counts,
    int32_t* __restrict__ rank) {
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; j < nnz; j += stride) {
    const int64_t key = keys[j];
    if (key == PAD_KEY) {
      inverse[j] = -1;
      rank[j] = 0;
      continue;
    }
    uint64_t h = mix_hash((uint64_t)key) & (uint64_t)cap_mask;
    for (int64_t probe = 0; probe <= cap_mask; ++probe) {
      int64_t idx = (int64_t)((h + probe) & (uint64_t)cap_mask);
      if (ht_keys[idx] == key) {
        int c = ht_compact[idx];
        inverse[j] = c;
        rank[j] = atomicAdd(&counts[c], 1);
        break;
      }
    }
  }
}

// slots_elem[j] = inverse[j] < 0 ? -1 : slots[inverse[j]]
// (replaces a .long() cast + index_select pair)
__global__ void k_slots_gather_pad(const int32_t* __restrict__ inverse,
                                   const int32_t* __restrict__ slots,
                                   int nnz, int32_t* __restrict__ out) {
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; j < nnz; j += stride) {
    int c = inverse[j];
    out[j] = c < 0 ? -1 : slots[c];
  }
}

// cnt_sum[inverse[j]] += cnt[j] for non-pad elements. A unique key
// repeats at most `world` times here (peers), so atomic contention is
// bounded by the world size.
__global__ void k_cnt_sum_pad(const int32_t* __restrict__ inverse,
                              const int32_t* __restrict__ cnt, int nnz,
                              int32_t* __restrict__ cnt_sum) {
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; j < nnz; j += stride) {
    int c = inverse[j];
    if (c >= 0) atomicAdd(&cnt_sum[c], cnt[j]);
  }
}

// grad2[inverse[j]] += grad[j] rows for non-pad elements (the owner-side
// cross-peer gradient reduction; <= world-way contention per row).
__global__ void k_rows_segsum_pad(const float* __restrict__ grad,
                                  const int32_t* __restrict__ inverse,
                                  int nnz, int dim,
                                  float* __restrict__ grad2) {
  int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t total = (int64_t)nnz * dim;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; t < total; t += stride) {
    int j = (int)(t / dim);
    int c = inverse[j];
    if (c < 0) continue;
    int d = (int)(t % dim);
    atomicAdd(&grad2[(int64_t)c * dim + d], grad[t]);
  }
}

// inv_out[j] = lut[inv_in[j]] — composes the requester inverse with the
// routed positions in one pass (no int64 cast round trip).
__global__ void k_compose_i32(const int32_t* __restrict__ inv_in,
                              const int32_t* __restrict__ lut, int nnz,
                              int32_t* __restrict__ inv_out) {
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; j < nnz; j += stride) inv_out[j] = lut[inv_in[j]];
}

// dst[route_pos[u]] = src[u] for u < m (grad rows into the padded wire
// layout; dst is pre-zeroed so pad rows carry zero gradient).
__global__ void k_rows_to_padded(const float* __restrict__ src,
                                 const int32_t* __restrict__ route_pos,
                                 const int32_t* __restrict__ m_dev,
                                 int n_cap, int dim,
                                 float* __restrict__ dst) {
  const int m = min(*m_dev, n_cap);
  int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t total = (int64_t)m * dim;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; t < total; t += stride) {
    int u = (int)(t / dim);
    int d = (int)(t % dim);
    dst[(int64_t)route_pos[u] * dim + d] = src[t];
  }
}

// Pass C (per occurrence): inverse + per-batch counts.
__global__ void k_dedup_pass_c(
    const int64_t* __restrict__ keys, int nnz,
    const int64_t* __restrict__ ht_keys,
    const int32_t* __restrict__ ht_compact, int64_t cap_mask,
    int32_t* __restrict__ inverse, int32_t* __restrict__ counts,
    int32_t* __restrict__ rank) {
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; j < nnz; j += stride) {
    const int64_t key = keys[j];
    uint64_t h = mix_hash((uint64_t)key) & (uint64_t)cap_mask;
    for (int64_t probe = 0; probe <= cap_mask; ++probe) {
      int64_t idx = (int64_t)((h + probe) & (uint64_t)cap_mask);
      if (ht_keys[idx] == key) {
        int c = ht_compact[idx];
        inverse[j] = c;
        // the counter value doubles as this occurrence's rank within its
        // key, letting the CSR order build become a direct scatter
        // (k_csr_scatter) instead of a second atomic-cursor pass
        rank[j] = atomicAdd(&counts[c], 1);
        break;
      }
    }
  }
}

// Index-direct pass C: pass A recorded each occurrence's hash entry, so
// inverse/rank need NO re-probe of the table (the probe was ~half of
// the dedup cost at 213k occurrences over a 16M-entry table).
__global__ void k_dedup_pass_c_idx(
    const int64_t* __restrict__ occ_entry, int nnz,
    const int32_t* __restrict__ ht_compact,
    int32_t* __restrict__ inverse, int32_t* __restrict__ counts,
    int32_t* __restrict__ rank) {
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; j < nnz; j += stride) {
    int c = ht_compact[occ_entry[j]];
    inverse[j] = c;
    rank[j] = atomicAdd(&counts[c], 1);
  }
}

// PAD-aware variant for the owner side of the sharded exchange.
__global__ void k_dedup_pass_c_idx_pad(
    const int64_t* __restrict__ keys,
    const int64_t* __restrict__ occ_entry, int nnz,
    const int32_t* __restrict__ ht_compact,
    int32_t* __restrict__ inverse, int32_t* __restrict__ counts,
    int32_t* __restrict__ rank) {
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; j < nnz; j += stride) {
    if (keys[j] == PAD_KEY) {
      inverse[j] = -1;
      rank[j] = 0;
      continue;
    }
    int c = ht_compact[occ_entry[j]];
    inverse[j] = c;
    rank[j] = atomicAdd(&counts[c], 1);
  }
}

// order[bounds[c] + rank[j]] = j — direct scatter using pass C's ranks.
// The destination is bounds-guarded: a corrupt inverse/rank pair (e.g. a
// key pass C failed to find) must trip the error flag, not fault the GPU.
__global__ void k_csr_scatter(const int32_t* __restrict__ inverse,
                              const int32_t* __restrict__ rank, int nnz,
                              const int32_t* __restrict__ bounds,
                              int32_t* __restrict__ order,
                              int32_t* __restrict__ error_flag) {
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; j < nnz; j += stride) {
    int64_t o = (int64_t)bounds[inverse[j]] + rank[j];
    if (o < 0 || o >= nnz) {
      atomicExch(error_flag, 3);
      continue;
    }
    order[o] = (int32_t)j;
  }
}

// CSR order build: order[bounds[c] + pos++] = j (pos via per-key cursor).
__global__ void k_csr_order(const int32_t* __restrict__ inverse, int nnz,
                            const int32_t* __restrict__ bounds,
                            int32_t* __restrict__ cursor,
                            int32_t* __restrict__ order) {
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; j < nnz; j += stride) {
    int c = inverse[j];
    int pos = atomicAdd(&cursor[c], 1);
    order[bounds[c] + pos] = (int32_t)j;
  }
}

// Zero-copy gather of cold-tier rows: the GPU dereferences pinned host
// memory directly (unified addressing), so the multi-tier staging path
// (≙ CopyEmbeddingsFromDramToHbm, hbm_dram_storage.h:412-435) is ONE
// kernel at interconnect bandwidth instead of a single-threaded CPU
// gather + copy (measured 0.4 GB/s on the CPU path).
__global__ void k_gather_host_rows(const float* __restrict__ host_src,
                                   const int64_t* __restrict__ rows,
                                   int64_t m, int dim,
                                   float* __restrict__ out) {
  int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t total = m * dim;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; t < total; t += stride)
    out[t] = host_src[rows[t / dim] * dim + t % dim];
}

// Reverse direction: scatter updated rows back into the pinned cold slab
// (cold-tier optimizer writes).
__global__ void k_scatter_host_rows(const float* __restrict__ src,
                                    const int64_t* __restrict__ rows,
                                    int64_t m, int dim,
                                    float* __restrict__ host_dst) {
  int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t total = m * dim;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; t < total; t += stride)
    host_dst[rows[t / dim] * dim + t % dim] = src[t];
}

// Read-only probe (serving / frequency / version queries).
// out_slots: slot or -1; out_entry: hash index or -1 (metadata access).
__global__ void k_lookup(
    const int64_t* __restrict__ keys, int n,
    const int64_t* __restrict__ ht_keys, const int32_t* __restrict__ ht_slot,
    int64_t cap_mask, int32_t* __restrict__ out_slots,
    int64_t* __restrict__ out_entry) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t gstride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += gstride) {
    const int64_t key = keys[i];
    uint64_t h = mix_hash((uint64_t)key) & (uint64_t)cap_mask;
    bool found = false;
    for (int64_t probe = 0; probe <= cap_mask; ++probe) {
      int64_t idx = (int64_t)((h + probe) & (uint64_t)cap_mask);
      int64_t cur = ht_keys[idx];
      if (cur == key) {
        out_slots[i] = ht_slot[idx];
        if (out_entry) out_entry[i] = idx;
        found = true;
        break;
      }
      if (cur == EMPTY_KEY) break;
    }
    if (!found) {
      out_slots[i] = -1;
      if (out_entry) out_entry[i] = -1;
    }
  }
}

// Export scan: compact live hash entries into dense output arrays.
__global__ void k_export_scan(
    const int64_t* __restrict__ ht_keys, const int32_t* __restrict__ ht_slot,
    const int32_t* __restrict__ ht_freq,
    const int64_t* __restrict__ ht_version, int64_t capacity,
    int32_t* __restrict__ cursor, int64_t* __restrict__ out_keys,
    int32_t* __restrict__ out_slots, int32_t* __restrict__ out_freqs,
    int64_t* __restrict__ out_versions, int out_cap) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < capacity; i += stride) {
    int64_t k = ht_keys[i];
    if (k == EMPTY_KEY) continue;
    int32_t pos = atomicAdd(cursor, 1);
    if (pos >= out_cap) continue;
    out_keys[pos] = k;
    out_slots[pos] = ht_slot[i];
    out_freqs[pos] = ht_freq[i];
    out_versions[pos] = ht_version[i];
  }
}

// ---------------------------------------------------------------------
// gather / pooled forward / pooled backward
// ---------------------------------------------------------------------

// Plain gather of unique rows: out[i] = values[slots[i]] or default row.
template <typename OutT>
__global__ void k_gather(
    const float* __restrict__ values, const float* __restrict__ default_values,
    const int64_t* __restrict__ keys, const int32_t* __restrict__ slots,
    int m, int dim, int default_value_dim, float no_permission_value,
    int use_no_permission, OutT* __restrict__ out) {
  int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t total = (int64_t)m * dim;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; t < total; t += stride) {
    int i = (int)(t / dim);
    int d = (int)(t % dim);
    int32_t s = slots[i];
    float v;
    if (s >= 0) {
      v = values[(int64_t)s * dim + d];
    } else if (use_no_permission) {
      v = no_permission_value;
    } else {
      int64_t row = (int64_t)((uint64_t)keys[i] % (uint64_t)default_value_dim);
      v = default_values[row * dim + d];
    }
    if constexpr (std::is_same_v<OutT, __hip_bfloat16>) out[t] = f2bf(v);
    else out[t] = v;
  }
}

DEV_INLINE float row_coeff(int combiner, int len, const float* wsum) {
  // combiner: 0=sum 1=mean 2=sqrtn; wsum = sum w (mean) / sum w^2 (sqrtn)
  if (combiner == 0 || len == 0) return 1.0f;
  float denom = wsum ? *wsum : (float)len;
  if (combiner == 2) denom = sqrtf(wsum ? *wsum : (float)len);
  return denom > 1e-12f ? 1.0f / denom : 0.0f;
}

// Fused gather + segment pooling. Thread = (batch row b, dim d); loops the
// row's ids accumulating values[slots[inverse[j]]][d]. No [nnz, dim] or
// [m, dim] intermediate is materialized.
template <typename OutT>
__global__ void k_pooled_fwd(
    const float* __restrict__ values, const float* __restrict__ default_values,
    const int64_t* __restrict__ keys, const int32_t* __restrict__ slots,
    const int32_t* __restrict__ inverse, const int32_t* __restrict__ offsets,
    const float* __restrict__ weights, int batch, int dim,
    int default_value_dim, float no_permission_value, int use_no_permission,
    int combiner, OutT* __restrict__ out) {
  int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t total = (int64_t)batch * dim;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; t < total; t += stride) {
    int b = (int)(t / dim);
    int d = (int)(t % dim);
    int beg = offsets[b], end = offsets[b + 1];
    float acc = 0.0f, wacc = 0.0f;
    for (int j = beg; j < end; ++j) {
      int u = inverse[j];
      int32_t s = slots[u];
      float v;
      if (s >= 0) {
        v = values[(int64_t)s * dim + d];
      } else if (use_no_permission) {
        v = no_permission_value;
      } else {
        int64_t row =
            (int64_t)((uint64_t)keys[u] % (uint64_t)default_value_dim);
        v = default_values[row * dim + d];
      }
      float w = weights ? weights[j] : 1.0f;
      acc += w * v;
      wacc += (combiner == 2) ? w * w : w;
    }
    float coeff = 1.0f;
    int len = end - beg;
    if (combiner != 0 && len > 0) {
      float denom = (combiner == 2) ? sqrtf(wacc) : wacc;
      coeff = denom > 1e-12f ? 1.0f / denom : 0.0f;
    }
    float r = acc * coeff;
    if constexpr (std::is_same_v<OutT, __hip_bfloat16>) out[t] = f2bf(r);
    else out[t] = r;
  }
}

// Pooled backward: grad_unique[u][d] += coeff(b) * w_j * grad_out[b][d]
// for every occurrence j of unique key u. Thread = (b, d); atomicAdd into
// grad_unique (unique keys repeat across rows).
template <typename GradT>
__global__ void k_pooled_bwd(
    const GradT* __restrict__ grad_out, const int32_t* __restrict__ inverse,
    const int32_t* __restrict__ offsets, const float* __restrict__ weights,
    int batch, int dim, int combiner, float* __restrict__ grad_unique) {
  int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t total = (int64_t)batch * dim;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; t < total; t += stride) {
    int b = (int)(t / dim);
    int d = (int)(t % dim);
    int beg = offsets[b], end = offsets[b + 1];
    int len = end - beg;
    if (len == 0) continue;
    float g;
    if constexpr (std::is_same_v<GradT, __hip_bfloat16>) g = bf2f(grad_out[t]);
    else g = grad_out[t];
    float coeff = 1.0f;
    if (combiner != 0) {
      float wacc = 0.0f;
      if (weights) {
        for (int j = beg; j < end; ++j)
          wacc += (combiner == 2) ? weights[j] * weights[j] : weights[j];
      } else {
        wacc = (float)len;
      }
      float denom = (combiner == 2) ? sqrtf(wacc) : wacc;
      coeff = denom > 1e-12f ? 1.0f / denom : 0.0f;
    }
    float gc = g * coeff;
    for (int j = beg; j < end; ++j) {
      float w = weights ? weights[j] : 1.0f;
      atomicAdd(&grad_unique[(int64_t)inverse[j] * dim + d], w * gc);
    }
  }
}

// ---------------------------------------------------------------------
// grouped (multi-table) fused pooling — the EmbeddingCollection hot path
// ---------------------------------------------------------------------
//
// N tables of equal dim share one storage via composite keys
// (table_id << key_bits) | id. One unique + one probe + one fused kernel
// per step regardless of table count (reference capability:
// GroupEmbeddingVarLookup, ops/kv_variable_ops.cc:404, re-designed so the
// whole group is ONE launch instead of one block-set per table).
//
// Ragged layout: the N tables' ragged batches are concatenated —
// offsets[(N*B)+1]; pooled row r (< N*B) is table t = r / B, sample
// b = r % B. Output is written interleaved as out[b, t*D + d] giving the
// [B, N*D] concat layout models consume with no extra copy.

// direct != nullptr (sharded path): embedding rows come pre-gathered per
// unique key (from peer shards over RCCL all-to-all) instead of the local
// slot slab.
template <typename OutT>
__global__ void k_group_pooled_fwd(
    const float* __restrict__ values, const float* __restrict__ default_values,
    const int64_t* __restrict__ keys, const int32_t* __restrict__ slots,
    const int32_t* __restrict__ inverse, const int32_t* __restrict__ offsets,
    const float* __restrict__ weights,
    const int32_t* __restrict__ combiner_ids, int batch, int n_tables,
    int dim, int default_value_dim, int key_bits, float no_permission_value,
    int use_no_permission, const float* __restrict__ direct,
    OutT* __restrict__ out) {
  int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t total = (int64_t)batch * n_tables * dim;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  const int64_t key_mask = ((int64_t)1 << key_bits) - 1;
  for (; t < total; t += stride) {
    int d = (int)(t % dim);
    int64_t rid = t / dim;           // pooled row in [0, N*B)
    int table = (int)(rid / batch);
    int b = (int)(rid % batch);
    int beg = offsets[rid], end = offsets[rid + 1];
    int combiner = combiner_ids[table];
    float acc = 0.0f, wacc = 0.0f;
    for (int j = beg; j < end; ++j) {
      int u = inverse[j];
      float v;
      if (direct) {
        v = direct[(int64_t)u * dim + d];
      } else {
        int32_t s = slots[u];
        if (s >= 0) {
          v = values[(int64_t)s * dim + d];
        } else if (use_no_permission) {
          v = no_permission_value;
        } else {
          int64_t k = keys[u];
          int64_t row = (int64_t)(table)*default_value_dim +
                        (int64_t)((uint64_t)(k & key_mask) %
                                  (uint64_t)default_value_dim);
          v = default_values[row * dim + d];
        }
      }
      float w = weights ? weights[j] : 1.0f;
      acc += w * v;
      wacc += (combiner == 2) ? w * w : w;
    }
    float coeff = 1.0f;
    if (combiner != 0 && end > beg) {
      float denom = (combiner == 2) ? sqrtf(wacc) : wacc;
      coeff = denom > 1e-12f ? 1.0f / denom : 0.0f;
    }
    float r = acc * coeff;
    int64_t oidx = ((int64_t)b * n_tables + table) * dim + d;
    if constexpr (std::is_same_v<OutT, __hip_bfloat16>) out[oidx] = f2bf(r);
    else out[oidx] = r;
  }
}

// IDENTITY: the matrix fast path (one id per table per sample) has
// row_ids == arange and unit combiner coefficients — skip both reads.
template <typename GradT, int SPLITS, int CHUNK, bool IDENTITY>
__global__ void k_group_pooled_bwd_strided(
    const GradT* __restrict__ grad_out, const int32_t* __restrict__ order,
    const int32_t* __restrict__ bounds, const int32_t* __restrict__ row_ids,
    const float* __restrict__ weights, const float* __restrict__ row_coeff,
    int m, const int32_t* __restrict__ m_dev, int batch, int n_tables,
    int dim, float* __restrict__ grad_unique) {
  if (m_dev) m = *m_dev;  // graph capture: true unique count on device
  int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t total = (int64_t)m * dim * SPLITS;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; t < total; t += stride) {
    int d = (int)(t % dim);
    int64_t ud = t / dim;
    int u = (int)(ud % m);
    int s = (int)(ud / m);
    int beg = bounds[u], end = bounds[u + 1];
    int cnt = end - beg;
    if (s * CHUNK >= cnt) continue;   // this split has no window
    float acc = 0.0f;
    for (int k0 = beg + s * CHUNK; k0 < end; k0 += SPLITS * CHUNK) {
      int k1 = min(k0 + CHUNK, end);
      for (int k = k0; k < k1; ++k) {
        int j = order[k];
        int rid = IDENTITY ? j : row_ids[j];
        int table = rid / batch;
        int b = rid % batch;
        float g;
        int64_t gidx = ((int64_t)b * n_tables + table) * dim + d;
        if constexpr (std::is_same_v<GradT, __hip_bfloat16>)
          g = bf2f(grad_out[gidx]);
        else
          g = grad_out[gidx];
        if constexpr (IDENTITY) {
          acc += g;
        } else {
          float w = weights ? weights[j] : 1.0f;
          acc += w * row_coeff[rid] * g;
        }
      }
    }
    int64_t o = (int64_t)u * dim + d;
    if (cnt <= CHUNK) grad_unique[o] = acc;   // single window: plain store
    else atomicAdd(&grad_unique[o], acc);
  }
}

// ---------------------------------------------------------------------
// fused sparse optimizer applies (thread = (unique key i, dim d))
// ---------------------------------------------------------------------

#define APPLY_PROLOG                                              \
  int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;     \
  int64_t total = (int64_t)m * dim;                               \
  int64_t stride = gridDim.x * (int64_t)blockDim.x;               \
  for (; t < total; t += stride) {                                \
    int i = (int)(t / dim);                                       \
    int d = (int)(t % dim);                                       \
    int32_t s = slots[i];                                         \
    if (s < 0) continue;                                          \
    int64_t o = (int64_t)s * dim + d;                             \
    float g = grad[t];

#define APPLY_EPILOG }

__global__ void k_apply_sgd(float* __restrict__ w,
                            const int32_t* __restrict__ slots,
                            const float* __restrict__ grad, int m, int dim,
                            float lr) {
  APPLY_PROLOG
  w[o] -= lr * g;
  APPLY_EPILOG
}

__global__ void k_apply_adagrad(float* __restrict__ w,
                                float* __restrict__ accum,
                                const int32_t* __restrict__ slots,
                                const float* __restrict__ grad, int m,
                                int dim, float lr, float epsilon) {
  APPLY_PROLOG
  float a = accum[o] + g * g;
  accum[o] = a;
  w[o] -= lr * g / (sqrtf(a) + epsilon);
  APPLY_EPILOG
}

// Periodic accumulator decay; period bookkeeping is per-slot (1-wide slab).
// Lane d==0 updates the period after all lanes read it — benign ordering:
// every thread computes the same decay from the pre-update period value
// because the period slab is read before any write (separate arrays).
__global__ void k_apply_adagrad_decay(
    float* __restrict__ w, float* __restrict__ accum,
    float* __restrict__ period_slab, const int32_t* __restrict__ slots,
    const float* __restrict__ grad, int m, int dim, float lr, float epsilon,
    float cur_period, float decay_rate, float baseline) {
  APPLY_PROLOG
  float prev_period = period_slab[s];
  float dp = cur_period - prev_period;
  if (dp < 0.0f) dp = 0.0f;
  float decay = powf(decay_rate, dp);
  float a = accum[o] * decay;
  if (a < baseline) a = baseline;
  a += g * g;
  accum[o] = a;
  w[o] -= lr * g / (sqrtf(a) + epsilon);
  APPLY_EPILOG
  // period_slab is committed by a separate k_commit_period launch so every
  // (key, d) thread here reads the pre-update period value.
}

__global__ void k_commit_period(float* __restrict__ period_slab,
                                const int32_t* __restrict__ slots, int m,
                                float cur_period) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t gstride = gridDim.x * (int64_t)blockDim.x;
  for (; i < m; i += gstride) {
    int32_t sl = slots[i];
    if (sl >= 0) period_slab[sl] = cur_period;
  }
}

__global__ void k_apply_adam(float* __restrict__ w, float* __restrict__ mom,
                             float* __restrict__ vel,
                             const int32_t* __restrict__ slots,
                             const float* __restrict__ grad, int m, int dim,
                             float lr_t, float beta1, float beta2,
                             float epsilon) {
  APPLY_PROLOG
  float mn = beta1 * mom[o] + (1.0f - beta1) * g;
  float vn = beta2 * vel[o] + (1.0f - beta2) * g * g;
  mom[o] = mn;
  vel[o] = vn;
  w[o] -= lr_t * mn / (sqrtf(vn) + epsilon);
  APPLY_EPILOG
}

__global__ void k_apply_adam_dev(float* __restrict__ w,
                                 float* __restrict__ mom,
                                 float* __restrict__ vel,
                                 const int32_t* __restrict__ slots,
                                 const float* __restrict__ grad, int m,
                                 int dim, float lr, float beta1, float beta2,
                                 float epsilon,
                                 const float* __restrict__ powers) {
  const float lr_t =
      lr * sqrtf(1.0f - powers[1]) / (1.0f - powers[0]);
  APPLY_PROLOG
  float mn = beta1 * mom[o] + (1.0f - beta1) * g;
  float vn = beta2 * vel[o] + (1.0f - beta2) * g * g;
  mom[o] = mn;
  vel[o] = vn;
  w[o] -= lr_t * mn / (sqrtf(vn) + epsilon);
  APPLY_EPILOG
}

__global__ void k_update_powers(float* __restrict__ powers, float beta1,
                                float beta2) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    powers[0] *= beta1;
    powers[1] *= beta2;
  }
}

__global__ void k_apply_adamw(float* __restrict__ w, float* __restrict__ mom,
                              float* __restrict__ vel,
                              const int32_t* __restrict__ slots,
                              const float* __restrict__ grad, int m, int dim,
                              float lr_t, float lr, float beta1, float beta2,
                              float epsilon, float weight_decay) {
  APPLY_PROLOG
  float mn = beta1 * mom[o] + (1.0f - beta1) * g;
  float vn = beta2 * vel[o] + (1.0f - beta2) * g * g;
  mom[o] = mn;
  vel[o] = vn;
  float wv = w[o];
  w[o] = wv - lr_t * mn / (sqrtf(vn) + epsilon) - lr * weight_decay * wv;
  APPLY_EPILOG
}

__global__ void k_apply_rmsprop(float* __restrict__ w, float* __restrict__ vel,
                                const int32_t* __restrict__ slots,
                                const float* __restrict__ grad, int m,
                                int dim, float lr, float beta2,
                                float epsilon) {
  APPLY_PROLOG
  float vn = beta2 * vel[o] + (1.0f - beta2) * g * g;
  vel[o] = vn;
  w[o] -= lr * g / (sqrtf(vn) + epsilon);
  APPLY_EPILOG
}

// l2_shrinkage != 0 is the FtrlV2 variant (reference:
// KvResourceSparseApplyFtrlV2): the linear term sees
// g + 2*l2_shrinkage*w while the accumulator sees plain g^2.
__global__ void k_apply_ftrl(float* __restrict__ w, float* __restrict__ n,
                             float* __restrict__ z,
                             const int32_t* __restrict__ slots,
                             const float* __restrict__ grad, int m, int dim,
                             float lr, float l1, float l2, float lr_power,
                             float l2_shrinkage) {
  APPLY_PROLOG
  float wv = w[o];
  float n_old = n[o];
  float n_new = n_old + g * g;
  float sigma = (powf(n_new, -lr_power) - powf(n_old, -lr_power)) / lr;
  float gs = g + 2.0f * l2_shrinkage * wv;
  float z_new = z[o] + gs - sigma * wv;
  n[o] = n_new;
  z[o] = z_new;
  float quad = powf(n_new, -lr_power) / lr + 2.0f * l2;
  float w_new = 0.0f;
  if (fabsf(z_new) > l1)
    w_new = ((z_new > 0.0f ? l1 : -l1) - z_new) / quad;
  w[o] = w_new;
  APPLY_EPILOG
}

}  // namespace

// ---------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------

#define CHECK_DEV(t) TORCH_CHECK((t).is_cuda(), #t " must be on GPU")

static inline int n_blocks(int64_t total, int block = kBlock) {
  int64_t b = (total + block - 1) / block;
  // keep the chip filled but bounded; grid-stride covers the tail
  return (int)std::min<int64_t>(b, 65535);
}

torch::Tensor ht_lookup_insert(
    torch::Tensor keys, torch::Tensor counts, torch::Tensor ht_keys,
    torch::Tensor ht_slot, torch::Tensor ht_freq, torch::Tensor ht_version,
    torch::Tensor slot_counter, torch::Tensor entry_counter,
    torch::Tensor values, torch::Tensor default_values, int64_t max_slots,
    int64_t dvd_per_table, int64_t key_bits, int64_t init_limit,
    int64_t filter_freq, int64_t step, bool train,
    torch::Tensor error_flag) {
  CHECK_DEV(keys);
  int n = keys.numel();
  auto out = torch::empty({n}, keys.options().dtype(torch::kInt32));
  if (n == 0) return out;
  auto stream = current_stream();
  int dim = values.size(1);
  k_lookup_insert<<<n_blocks(n), kBlock, 0, stream>>>(
      keys.data_ptr<int64_t>(),
      counts.defined() && counts.numel() ? counts.data_ptr<int32_t>()
                                         : nullptr,
      n, ht_keys.data_ptr<int64_t>(), ht_slot.data_ptr<int32_t>(),
      ht_freq.data_ptr<int32_t>(), ht_version.data_ptr<int64_t>(),
      ht_keys.numel() - 1, slot_counter.data_ptr<int32_t>(),
      entry_counter.data_ptr<int32_t>(), (int)max_slots,
      values.data_ptr<float>(), default_values.data_ptr<float>(), dim,
      (int)dvd_per_table, (int)key_bits, (int)init_limit, (int)filter_freq,
      step, train ? 1 : 0, out.data_ptr<int32_t>(),
      error_flag.data_ptr<int32_t>());
  return out;
}

torch::Tensor ht_dedup_a(torch::Tensor keys, torch::Tensor ht_keys,
                         torch::Tensor ht_freq, torch::Tensor ht_version,
                         torch::Tensor ht_epoch, torch::Tensor ht_compact,
                         int64_t epoch, int64_t step,
                         torch::Tensor entry_counter, torch::Tensor m_counter,
                         torch::Tensor uniq_keys, torch::Tensor compact_entry,
                         torch::Tensor occ_entry, torch::Tensor error_flag) {
  int64_t nnz = keys.numel();
  if (nnz == 0) return m_counter;
  auto stream = current_stream();
  k_dedup_pass_a<<<n_blocks(nnz), kBlock, 0, stream>>>(
      keys.data_ptr<int64_t>(), (int)nnz, ht_keys.data_ptr<int64_t>(),
      ht_freq.data_ptr<int32_t>(), ht_version.data_ptr<int64_t>(),
      ht_epoch.data_ptr<int32_t>(), ht_compact.data_ptr<int32_t>(),
      ht_keys.numel() - 1, (int)epoch, step,
      entry_counter.data_ptr<int32_t>(), m_counter.data_ptr<int32_t>(),
      uniq_keys.data_ptr<int64_t>(), compact_entry.data_ptr<int64_t>(),
      occ_entry.data_ptr<int64_t>(), error_flag.data_ptr<int32_t>());
  return m_counter;
}

torch::Tensor ht_dedup_b(torch::Tensor compact_entry, torch::Tensor uniq_keys,
                         torch::Tensor ht_slot, torch::Tensor ht_freq,
                         torch::Tensor ht_version,
                         torch::Tensor batch_counts, int64_t step,
                         torch::Tensor slot_counter, int64_t max_slots,
                         torch::Tensor values, torch::Tensor default_values,
                         int64_t dvd_per_table, int64_t key_bits,
                         int64_t init_limit, int64_t filter_freq,
                         torch::Tensor error_flag) {
  int m = uniq_keys.numel();
  auto slots = torch::empty({m}, ht_slot.options());
  if (m == 0) return slots;
  auto stream = current_stream();
  k_dedup_pass_b<<<n_blocks(m), kBlock, 0, stream>>>(
      compact_entry.data_ptr<int64_t>(), uniq_keys.data_ptr<int64_t>(), m,
      ht_slot.data_ptr<int32_t>(), ht_freq.data_ptr<int32_t>(),
      ht_version.data_ptr<int64_t>(), batch_counts.data_ptr<int32_t>(),
      step, slot_counter.data_ptr<int32_t>(), (int)max_slots,
      values.data_ptr<float>(), default_values.data_ptr<float>(),
      values.size(1), (int)dvd_per_table, (int)key_bits, (int)init_limit,
      (int)filter_freq, slots.data_ptr<int32_t>(),
      error_flag.data_ptr<int32_t>());
  return slots;
}

void bump_epoch(torch::Tensor epoch_dev, torch::Tensor step_dev) {
  k_bump_epoch<<<1, 64, 0, current_stream()>>>(
      epoch_dev.data_ptr<int32_t>(), step_dev.data_ptr<int64_t>());
}

void bump_epoch_only(torch::Tensor epoch_dev) {
  k_bump_epoch_only<<<1, 64, 0, current_stream()>>>(
      epoch_dev.data_ptr<int32_t>());
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> route_pad(
    torch::Tensor uniq_keys, torch::Tensor counts, torch::Tensor m_dev,
    int64_t world, int64_t cap, int64_t key_bits,
    torch::Tensor error_flag) {
  int n_cap = uniq_keys.numel();
  int total = (int)(world * cap);
  auto opts_i32 = counts.options();
  auto send_keys = torch::empty({(int64_t)total}, uniq_keys.options());
  auto send_cnt = torch::empty({(int64_t)total}, opts_i32);
  auto route_pos = torch::empty({(int64_t)n_cap}, opts_i32);
  auto cursor = torch::empty({world}, opts_i32);
  auto stream = current_stream();
  k_route_fill<<<n_blocks(total), kBlock, 0, stream>>>(
      send_keys.data_ptr<int64_t>(), send_cnt.data_ptr<int32_t>(),
      cursor.data_ptr<int32_t>(), total, (int)world);
  k_route_pad<<<n_blocks(n_cap), kBlock, 0, stream>>>(
      uniq_keys.data_ptr<int64_t>(), counts.data_ptr<int32_t>(),
      m_dev.data_ptr<int32_t>(), n_cap, (int)world, (int)cap, (int)key_bits,
      send_keys.data_ptr<int64_t>(), send_cnt.data_ptr<int32_t>(),
      route_pos.data_ptr<int32_t>(), cursor.data_ptr<int32_t>(),
      error_flag.data_ptr<int32_t>());
  return {send_keys, send_cnt, route_pos};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> ht_dedup_c_pad(
    torch::Tensor keys, torch::Tensor ht_keys, torch::Tensor ht_compact,
    int64_t m) {
  int64_t nnz = keys.numel();
  auto inverse = torch::empty({nnz}, ht_compact.options());
  auto rank = torch::empty({nnz}, ht_compact.options());
  auto counts = torch::empty({m}, ht_compact.options());
  if (nnz == 0) return {inverse, counts, rank};
  auto stream = current_stream();
  k_zero_f32<<<n_blocks((m + 3) / 4), kBlock, 0, stream>>>(
      reinterpret_cast<float*>(counts.data_ptr<int32_t>()), m);
  k_dedup_pass_c_pad<<<n_blocks(nnz), kBlock, 0, stream>>>(
      keys.data_ptr<int64_t>(), (int)nnz, ht_keys.data_ptr<int64_t>(),
      ht_compact.data_ptr<int32_t>(), ht_keys.numel() - 1,
      inverse.data_ptr<int32_t>(), counts.data_ptr<int32_t>(),
      rank.data_ptr<int32_t>());
  return {inverse, counts, rank};
}

torch::Tensor slots_gather_pad(torch::Tensor inverse, torch::Tensor slots) {
  int64_t nnz = inverse.numel();
  auto out = torch::empty({nnz}, slots.options());
  if (nnz == 0) return out;
  k_slots_gather_pad<<<n_blocks(nnz), kBlock, 0, current_stream()>>>(
      inverse.data_ptr<int32_t>(), slots.data_ptr<int32_t>(), (int)nnz,
      out.data_ptr<int32_t>());
  return out;
}

torch::Tensor cnt_sum_pad(torch::Tensor inverse, torch::Tensor cnt,
                          int64_t m_cap) {
  int64_t nnz = inverse.numel();
  auto out = torch::empty({m_cap}, cnt.options());
  auto stream = current_stream();
  k_zero_f32<<<n_blocks((m_cap + 3) / 4), kBlock, 0, stream>>>(
      reinterpret_cast<float*>(out.data_ptr<int32_t>()), m_cap);
  if (nnz)
    k_cnt_sum_pad<<<n_blocks(nnz), kBlock, 0, stream>>>(
        inverse.data_ptr<int32_t>(), cnt.data_ptr<int32_t>(), (int)nnz,
        out.data_ptr<int32_t>());
  return out;
}

torch::Tensor rows_segsum_pad(torch::Tensor grad, torch::Tensor inverse,
                              int64_t m_cap) {
  int64_t nnz = inverse.numel();
  int dim = grad.size(1);
  auto out = torch::empty({m_cap, (int64_t)dim}, grad.options());
  auto stream = current_stream();
  k_zero_f32<<<n_blocks((m_cap * dim + 3) / 4), kBlock, 0, stream>>>(
      out.data_ptr<float>(), m_cap * dim);
  if (nnz)
    k_rows_segsum_pad<<<n_blocks(nnz * dim), kBlock, 0, stream>>>(
        grad.data_ptr<float>(), inverse.data_ptr<int32_t>(), (int)nnz, dim,
        out.data_ptr<float>());
  return out;
}

torch::Tensor gather_host_rows(torch::Tensor host_src, torch::Tensor rows) {
  TORCH_CHECK(host_src.is_pinned(), "cold slab must be pinned host memory");
  TORCH_CHECK(rows.is_cuda() && rows.scalar_type() == torch::kInt64);
  int64_t m = rows.numel();
  int dim = host_src.size(1);
  auto out = torch::empty({m, (int64_t)dim},
                          rows.options().dtype(torch::kFloat32));
  if (m == 0) return out;
  k_gather_host_rows<<<n_blocks(m * dim), kBlock, 0, current_stream()>>>(
      host_src.data_ptr<float>(), rows.data_ptr<int64_t>(), m, dim,
      out.data_ptr<float>());
  return out;
}

void scatter_host_rows(torch::Tensor src, torch::Tensor rows,
                       torch::Tensor host_dst) {
  TORCH_CHECK(host_dst.is_pinned(), "cold slab must be pinned host memory");
  TORCH_CHECK(rows.is_cuda() && src.is_cuda());
  int64_t m = rows.numel();
  int dim = host_dst.size(1);
  if (m == 0) return;
  k_scatter_host_rows<<<n_blocks(m * dim), kBlock, 0, current_stream()>>>(
      src.data_ptr<float>(), rows.data_ptr<int64_t>(), m, dim,
      host_dst.data_ptr<float>());
}

torch::Tensor compose_i32(torch::Tensor inv_in, torch::Tensor lut) {
  int64_t nnz = inv_in.numel();
  auto out = torch::empty({nnz}, inv_in.options());
  if (nnz == 0) return out;
  k_compose_i32<<<n_blocks(nnz), kBlock, 0, current_stream()>>>(
      inv_in.data_ptr<int32_t>(), lut.data_ptr<int32_t>(), (int)nnz,
      out.data_ptr<int32_t>());
  return out;
}

torch::Tensor rows_to_padded(torch::Tensor src, torch::Tensor route_pos,
                             torch::Tensor m_dev, int64_t out_rows) {
  int n_cap = route_pos.numel();
  int dim = src.size(1);
  auto dst = torch::empty({out_rows, (int64_t)dim}, src.options());
  auto stream = current_stream();
  k_zero_f32<<<n_blocks((out_rows * dim + 3) / 4), kBlock, 0, stream>>>(
      dst.data_ptr<float>(), out_rows * dim);
  k_rows_to_padded<<<n_blocks((int64_t)n_cap * dim), kBlock, 0, stream>>>(
      src.data_ptr<float>(), route_pos.data_ptr<int32_t>(),
      m_dev.data_ptr<int32_t>(), n_cap, dim, dst.data_ptr<float>());
  return dst;
}

torch::Tensor ht_dedup_a_dev(
    torch::Tensor keys, torch::Tensor ht_keys, torch::Tensor ht_freq,
    torch::Tensor ht_version, torch::Tensor ht_epoch,
    torch::Tensor ht_compact, torch::Tensor epoch_dev,
    torch::Tensor step_dev, torch::Tensor entry_counter,
    torch::Tensor m_counter, torch::Tensor uniq_keys,
    torch::Tensor compact_entry, torch::Tensor occ_entry,
    torch::Tensor error_flag) {
  int64_t nnz = keys.numel();
  if (nnz == 0) return m_counter;
  auto stream = current_stream();
  k_dedup_pass_a_dev<<<n_blocks(nnz), kBlock, 0, stream>>>(
      keys.data_ptr<int64_t>(), (int)nnz, ht_keys.data_ptr<int64_t>(),
      ht_freq.data_ptr<int32_t>(), ht_version.data_ptr<int64_t>(),
      ht_epoch.data_ptr<int32_t>(), ht_compact.data_ptr<int32_t>(),
      ht_keys.numel() - 1, epoch_dev.data_ptr<int32_t>(),
      step_dev.data_ptr<int64_t>(), entry_counter.data_ptr<int32_t>(),
      m_counter.data_ptr<int32_t>(), uniq_keys.data_ptr<int64_t>(),
      compact_entry.data_ptr<int64_t>(), occ_entry.data_ptr<int64_t>(),
      error_flag.data_ptr<int32_t>());
  return m_counter;
}

torch::Tensor ht_dedup_b_padded(
    torch::Tensor compact_entry, torch::Tensor uniq_keys,
    torch::Tensor m_dev, torch::Tensor ht_slot, torch::Tensor ht_freq,
    torch::Tensor ht_version, torch::Tensor batch_counts,
    torch::Tensor step_dev,
    torch::Tensor slot_counter, int64_t max_slots, torch::Tensor values,
    torch::Tensor default_values, int64_t dvd_per_table, int64_t key_bits,
    int64_t init_limit, int64_t filter_freq, torch::Tensor error_flag) {
  int n_cap = uniq_keys.numel();
  auto slots = torch::empty({n_cap}, ht_slot.options());
  if (n_cap == 0) return slots;
  auto stream = current_stream();
  k_dedup_pass_b_padded<<<n_blocks(n_cap), kBlock, 0, stream>>>(
      compact_entry.data_ptr<int64_t>(), uniq_keys.data_ptr<int64_t>(),
      n_cap, m_dev.data_ptr<int32_t>(), ht_slot.data_ptr<int32_t>(),
      ht_freq.data_ptr<int32_t>(), ht_version.data_ptr<int64_t>(),
      batch_counts.data_ptr<int32_t>(), step_dev.data_ptr<int64_t>(),
      slot_counter.data_ptr<int32_t>(),
      (int)max_slots, values.data_ptr<float>(),
      default_values.data_ptr<float>(), values.size(1), (int)dvd_per_table,
      (int)key_bits, (int)init_limit, (int)filter_freq,
      slots.data_ptr<int32_t>(), error_flag.data_ptr<int32_t>());
  return slots;
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> ht_dedup_c(
    torch::Tensor keys, torch::Tensor ht_keys, torch::Tensor ht_compact,
    int64_t m) {
  int64_t nnz = keys.numel();
  auto inverse = torch::empty({nnz}, ht_compact.options());
  auto rank = torch::empty({nnz}, ht_compact.options());
  auto counts = torch::empty({m}, ht_compact.options());
  if (nnz == 0) return {inverse, counts, rank};
  auto stream = current_stream();
  k_zero_f32<<<n_blocks((m + 3) / 4), kBlock, 0, stream>>>(
      reinterpret_cast<float*>(counts.data_ptr<int32_t>()), m);
  k_dedup_pass_c<<<n_blocks(nnz), kBlock, 0, stream>>>(
      keys.data_ptr<int64_t>(), (int)nnz, ht_keys.data_ptr<int64_t>(),
      ht_compact.data_ptr<int32_t>(), ht_keys.numel() - 1,
      inverse.data_ptr<int32_t>(), counts.data_ptr<int32_t>(),
      rank.data_ptr<int32_t>());
  return {inverse, counts, rank};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> ht_dedup_c_idx(
    torch::Tensor occ_entry, torch::Tensor ht_compact, int64_t m) {
  int64_t nnz = occ_entry.numel();
  auto inverse = torch::empty({nnz}, ht_compact.options());
  auto rank = torch::empty({nnz}, ht_compact.options());
  auto counts = torch::empty({m}, ht_compact.options());
  if (nnz == 0) return {inverse, counts, rank};
  auto stream = current_stream();
  k_zero_f32<<<n_blocks((m + 3) / 4), kBlock, 0, stream>>>(
      reinterpret_cast<float*>(counts.data_ptr<int32_t>()), m);
  k_dedup_pass_c_idx<<<n_blocks(nnz), kBlock, 0, stream>>>(
      occ_entry.data_ptr<int64_t>(), (int)nnz,
      ht_compact.data_ptr<int32_t>(), inverse.data_ptr<int32_t>(),
      counts.data_ptr<int32_t>(), rank.data_ptr<int32_t>());
  return {inverse, counts, rank};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> ht_dedup_c_idx_pad(
    torch::Tensor keys, torch::Tensor occ_entry, torch::Tensor ht_compact,
    int64_t m) {
  int64_t nnz = occ_entry.numel();
  auto inverse = torch::empty({nnz}, ht_compact.options());
  auto rank = torch::empty({nnz}, ht_compact.options());
  auto counts = torch::empty({m}, ht_compact.options());
  if (nnz == 0) return {inverse, counts, rank};
  auto stream = current_stream();
  k_zero_f32<<<n_blocks((m + 3) / 4), kBlock, 0, stream>>>(
      reinterpret_cast<float*>(counts.data_ptr<int32_t>()), m);
  k_dedup_pass_c_idx_pad<<<n_blocks(nnz), kBlock, 0, stream>>>(
      keys.data_ptr<int64_t>(), occ_entry.data_ptr<int64_t>(), (int)nnz,
      ht_compact.data_ptr<int32_t>(), inverse.data_ptr<int32_t>(),
      counts.data_ptr<int32_t>(), rank.data_ptr<int32_t>());
  return {inverse, counts, rank};
}

torch::Tensor csr_scatter(torch::Tensor inverse, torch::Tensor rank,
                          torch::Tensor bounds, torch::Tensor error_flag) {
  int64_t nnz = inverse.numel();
  auto order = torch::empty({nnz}, inverse.options());
  if (nnz == 0) return order;
  k_csr_scatter<<<n_blocks(nnz), kBlock, 0, current_stream()>>>(
      inverse.data_ptr<int32_t>(), rank.data_ptr<int32_t>(), (int)nnz,
      bounds.data_ptr<int32_t>(), order.data_ptr<int32_t>(),
      error_flag.data_ptr<int32_t>());
  return order;
}

torch::Tensor csr_order(torch::Tensor inverse, torch::Tensor bounds,
                        int64_t m) {
  int64_t nnz = inverse.numel();
  auto order = torch::empty({nnz}, inverse.options());
  auto cursor = torch::zeros({m}, inverse.options());
  if (nnz == 0) return order;
  auto stream = current_stream();
  k_csr_order<<<n_blocks(nnz), kBlock, 0, stream>>>(
      inverse.data_ptr<int32_t>(), (int)nnz, bounds.data_ptr<int32_t>(),
      cursor.data_ptr<int32_t>(), order.data_ptr<int32_t>());
  return order;
}

void ht_insert_bulk(torch::Tensor keys, torch::Tensor slots,
                    torch::Tensor freqs, torch::Tensor versions,
                    torch::Tensor ht_keys, torch::Tensor ht_slot,
                    torch::Tensor ht_freq, torch::Tensor ht_version,
                    torch::Tensor entry_counter, torch::Tensor error_flag) {
  int n = keys.numel();
  if (n == 0) return;
  auto stream = current_stream();
  k_insert_bulk<<<n_blocks(n), kBlock, 0, stream>>>(
      keys.data_ptr<int64_t>(), slots.data_ptr<int32_t>(),
      freqs.defined() && freqs.numel() ? freqs.data_ptr<int32_t>() : nullptr,
      versions.defined() && versions.numel() ? versions.data_ptr<int64_t>()
                                             : nullptr,
      n, ht_keys.data_ptr<int64_t>(), ht_slot.data_ptr<int32_t>(),
      ht_freq.data_ptr<int32_t>(), ht_version.data_ptr<int64_t>(),
      ht_keys.numel() - 1, entry_counter.data_ptr<int32_t>(),
      error_flag.data_ptr<int32_t>());
}

std::tuple<torch::Tensor, torch::Tensor> ht_lookup(torch::Tensor keys,
                                                   torch::Tensor ht_keys,
                                                   torch::Tensor ht_slot,
                                                   bool want_entry) {
  int n = keys.numel();
  auto slots = torch::empty({n}, keys.options().dtype(torch::kInt32));
  auto entry = want_entry
                   ? torch::empty({n}, keys.options().dtype(torch::kInt64))
                   : torch::Tensor();
  if (n == 0) return {slots, entry};
  auto stream = current_stream();
  k_lookup<<<n_blocks(n), kBlock, 0, stream>>>(
      keys.data_ptr<int64_t>(), n, ht_keys.data_ptr<int64_t>(),
      ht_slot.data_ptr<int32_t>(), ht_keys.numel() - 1,
      slots.data_ptr<int32_t>(),
      want_entry ? entry.data_ptr<int64_t>() : nullptr);
  return {slots, entry};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
ht_export(torch::Tensor ht_keys, torch::Tensor ht_slot, torch::Tensor ht_freq,
          torch::Tensor ht_version, int64_t n_entries) {
  auto opts_i64 = ht_keys.options();
  auto opts_i32 = ht_slot.options();
  auto out_keys = torch::empty({n_entries}, opts_i64);
  auto out_slots = torch::empty({n_entries}, opts_i32);
  auto out_freqs = torch::empty({n_entries}, opts_i32);
  auto out_versions = torch::empty({n_entries}, opts_i64);
  auto cursor = torch::zeros({1}, opts_i32);
  if (n_entries > 0) {
    auto stream = current_stream();
    k_export_scan<<<n_blocks(ht_keys.numel()), kBlock, 0, stream>>>(
        ht_keys.data_ptr<int64_t>(), ht_slot.data_ptr<int32_t>(),
        ht_freq.data_ptr<int32_t>(), ht_version.data_ptr<int64_t>(),
        ht_keys.numel(), cursor.data_ptr<int32_t>(),
        out_keys.data_ptr<int64_t>(), out_slots.data_ptr<int32_t>(),
        out_freqs.data_ptr<int32_t>(), out_versions.data_ptr<int64_t>(),
        (int)n_entries);
  }
  return {out_keys, out_slots, out_freqs, out_versions};
}

torch::Tensor ev_gather(torch::Tensor values, torch::Tensor default_values,
                        torch::Tensor keys, torch::Tensor slots,
                        double no_permission_value, bool use_no_permission,
                        torch::ScalarType out_dtype) {
  int m = keys.numel();
  int dim = values.size(1);
  auto out = torch::empty({m, dim}, values.options().dtype(out_dtype));
  if (m == 0) return out;
  auto stream = current_stream();
  int64_t total = (int64_t)m * dim;
  if (out_dtype == torch::kBFloat16) {
    k_gather<__hip_bfloat16><<<n_blocks(total), kBlock, 0, stream>>>(
        values.data_ptr<float>(), default_values.data_ptr<float>(),
        keys.data_ptr<int64_t>(), slots.data_ptr<int32_t>(), m, dim,
        default_values.size(0), (float)no_permission_value,
        use_no_permission ? 1 : 0,
        reinterpret_cast<__hip_bfloat16*>(out.data_ptr<at::BFloat16>()));
  } else {
    k_gather<float><<<n_blocks(total), kBlock, 0, stream>>>(
        values.data_ptr<float>(), default_values.data_ptr<float>(),
        keys.data_ptr<int64_t>(), slots.data_ptr<int32_t>(), m, dim,
        default_values.size(0), (float)no_permission_value,
        use_no_permission ? 1 : 0, out.data_ptr<float>());
  }
  return out;
}

torch::Tensor pooled_fwd(torch::Tensor values, torch::Tensor default_values,
                         torch::Tensor keys, torch::Tensor slots,
                         torch::Tensor inverse, torch::Tensor offsets,
                         torch::Tensor weights, int64_t combiner,
                         double no_permission_value, bool use_no_permission,
                         torch::ScalarType out_dtype) {
  int batch = offsets.numel() - 1;
  int dim = values.size(1);
  auto out = torch::empty({batch, dim}, values.options().dtype(out_dtype));
  if (batch == 0) return out;
  auto stream = current_stream();
  int64_t total = (int64_t)batch * dim;
  const float* wptr =
      weights.defined() && weights.numel() ? weights.data_ptr<float>()
                                           : nullptr;
  if (out_dtype == torch::kBFloat16) {
    k_pooled_fwd<__hip_bfloat16><<<n_blocks(total), kBlock, 0, stream>>>(
        values.data_ptr<float>(), default_values.data_ptr<float>(),
        keys.data_ptr<int64_t>(), slots.data_ptr<int32_t>(),
        inverse.data_ptr<int32_t>(), offsets.data_ptr<int32_t>(), wptr, batch,
        dim, default_values.size(0), (float)no_permission_value,
        use_no_permission ? 1 : 0, (int)combiner,
        reinterpret_cast<__hip_bfloat16*>(out.data_ptr<at::BFloat16>()));
  } else {
    k_pooled_fwd<float><<<n_blocks(total), kBlock, 0, stream>>>(
        values.data_ptr<float>(), default_values.data_ptr<float>(),
        keys.data_ptr<int64_t>(), slots.data_ptr<int32_t>(),
        inverse.data_ptr<int32_t>(), offsets.data_ptr<int32_t>(), wptr, batch,
        dim, default_values.size(0), (float)no_permission_value,
        use_no_permission ? 1 : 0, (int)combiner, out.data_ptr<float>());
  }
  return out;
}

torch::Tensor pooled_bwd(torch::Tensor grad_out, torch::Tensor inverse,
                         torch::Tensor offsets, torch::Tensor weights,
                         int64_t m, int64_t combiner) {
  int batch = offsets.numel() - 1;
  int dim = grad_out.size(1);
  auto grad_unique = torch::zeros(
      {m, dim}, grad_out.options().dtype(torch::kFloat32));
  if (batch == 0 || m == 0) return grad_unique;
  auto stream = current_stream();
  int64_t total = (int64_t)batch * dim;
  const float* wptr =
      weights.defined() && weights.numel() ? weights.data_ptr<float>()
                                           : nullptr;
  if (grad_out.scalar_type() == torch::kBFloat16) {
    k_pooled_bwd<__hip_bfloat16><<<n_blocks(total), kBlock, 0, stream>>>(
        reinterpret_cast<const __hip_bfloat16*>(
            grad_out.data_ptr<at::BFloat16>()),
        inverse.data_ptr<int32_t>(), offsets.data_ptr<int32_t>(), wptr, batch,
        dim, (int)combiner, grad_unique.data_ptr<float>());
  } else {
    k_pooled_bwd<float><<<n_blocks(total), kBlock, 0, stream>>>(
        grad_out.data_ptr<float>(), inverse.data_ptr<int32_t>(),
        offsets.data_ptr<int32_t>(), wptr, batch, dim, (int)combiner,
        grad_unique.data_ptr<float>());
  }
  return grad_unique;
}

torch::Tensor group_pooled_fwd(
    torch::Tensor values, torch::Tensor default_values, torch::Tensor keys,
    torch::Tensor slots, torch::Tensor inverse, torch::Tensor offsets,
    torch::Tensor weights, torch::Tensor combiner_ids, int64_t batch,
    int64_t n_tables, int64_t key_bits, double no_permission_value,
    bool use_no_permission, torch::ScalarType out_dtype) {
  int dim = values.size(1);
  auto out = torch::empty({batch, n_tables * dim},
                          values.options().dtype(out_dtype));
  int64_t total = batch * n_tables * dim;
  if (total == 0) return out;
  auto stream = current_stream();
  const float* wptr =
      weights.defined() && weights.numel() ? weights.data_ptr<float>()
                                           : nullptr;
  if (out_dtype == torch::kBFloat16) {
    k_group_pooled_fwd<__hip_bfloat16>
        <<<n_blocks(total), kBlock, 0, stream>>>(
            values.data_ptr<float>(), default_values.data_ptr<float>(),
            keys.data_ptr<int64_t>(), slots.data_ptr<int32_t>(),
            inverse.data_ptr<int32_t>(), offsets.data_ptr<int32_t>(), wptr,
            combiner_ids.data_ptr<int32_t>(), (int)batch, (int)n_tables, dim,
            default_values.size(0) / (int)n_tables, (int)key_bits,
            (float)no_permission_value, use_no_permission ? 1 : 0, nullptr,
            reinterpret_cast<__hip_bfloat16*>(out.data_ptr<at::BFloat16>()));
  } else {
    k_group_pooled_fwd<float><<<n_blocks(total), kBlock, 0, stream>>>(
        values.data_ptr<float>(), default_values.data_ptr<float>(),
        keys.data_ptr<int64_t>(), slots.data_ptr<int32_t>(),
        inverse.data_ptr<int32_t>(), offsets.data_ptr<int32_t>(), wptr,
        combiner_ids.data_ptr<int32_t>(), (int)batch, (int)n_tables, dim,
        default_values.size(0) / (int)n_tables, (int)key_bits,
        (float)no_permission_value, use_no_permission ? 1 : 0, nullptr,
        out.data_ptr<float>());
  }
  return out;
}

torch::Tensor group_pooled_fwd_direct(
    torch::Tensor emb_rows, torch::Tensor keys, torch::Tensor inverse,
    torch::Tensor offsets, torch::Tensor weights, torch::Tensor combiner_ids,
    int64_t batch, int64_t n_tables, torch::ScalarType out_dtype) {
  int dim = emb_rows.size(1);
  auto out = torch::empty({batch, n_tables * dim},
                          emb_rows.options().dtype(out_dtype));
  int64_t total = batch * n_tables * dim;
  if (total == 0) return out;
  auto stream = current_stream();
  const float* wptr =
      weights.defined() && weights.numel() ? weights.data_ptr<float>()
                                           : nullptr;
  if (out_dtype == torch::kBFloat16) {
    k_group_pooled_fwd<__hip_bfloat16>
        <<<n_blocks(total), kBlock, 0, stream>>>(
            nullptr, nullptr, keys.data_ptr<int64_t>(), nullptr,
            inverse.data_ptr<int32_t>(), offsets.data_ptr<int32_t>(), wptr,
            combiner_ids.data_ptr<int32_t>(), (int)batch, (int)n_tables, dim,
            1, 0, 0.0f, 0, emb_rows.data_ptr<float>(),
            reinterpret_cast<__hip_bfloat16*>(out.data_ptr<at::BFloat16>()));
  } else {
    k_group_pooled_fwd<float><<<n_blocks(total), kBlock, 0, stream>>>(
        nullptr, nullptr, keys.data_ptr<int64_t>(), nullptr,
        inverse.data_ptr<int32_t>(), offsets.data_ptr<int32_t>(), wptr,
        combiner_ids.data_ptr<int32_t>(), (int)batch, (int)n_tables, dim, 1,
        0, 0.0f, 0, emb_rows.data_ptr<float>(), out.data_ptr<float>());
  }
  return out;
}

template <typename T, int SPL, int CHUNK = 128>
static void launch_bwd_strided(const T* gp, torch::Tensor& order,
                               torch::Tensor& bounds, torch::Tensor& row_ids,
                               const float* wptr, torch::Tensor& row_coeff,
                               int64_t m, const int32_t* mdp, int64_t batch,
                               int64_t n_tables, int64_t dim,
                               bool identity_rows, torch::Tensor& out,
                               hipStream_t stream) {
  // grid-stride kernel: cap the grid so an nnz-padded m (graph capture)
  // costs no extra block launches
  int64_t total = std::min<int64_t>(m * dim * SPL, (int64_t)4096 * kBlock);
  if (identity_rows) {
    k_group_pooled_bwd_strided<T, SPL, CHUNK, true>
        <<<n_blocks(total), kBlock, 0, stream>>>(
            gp, order.data_ptr<int32_t>(), bounds.data_ptr<int32_t>(),
            row_ids.data_ptr<int32_t>(), wptr, row_coeff.data_ptr<float>(),
            (int)m, mdp, (int)batch, (int)n_tables, (int)dim,
            out.data_ptr<float>());
  } else {
    k_group_pooled_bwd_strided<T, SPL, CHUNK, false>
        <<<n_blocks(total), kBlock, 0, stream>>>(
            gp, order.data_ptr<int32_t>(), bounds.data_ptr<int32_t>(),
            row_ids.data_ptr<int32_t>(), wptr, row_coeff.data_ptr<float>(),
            (int)m, mdp, (int)batch, (int)n_tables, (int)dim,
            out.data_ptr<float>());
  }
}

torch::Tensor group_pooled_bwd_strided(
    torch::Tensor grad_out, torch::Tensor order, torch::Tensor bounds,
    torch::Tensor row_ids, torch::Tensor weights, torch::Tensor row_coeff,
    int64_t m, torch::Tensor m_dev, int64_t batch, int64_t n_tables,
    int64_t dim, bool identity_rows, int64_t splits) {
  // splits: 8 for duplication-heavy (zipf) batches — hot keys fan out
  // across concurrent accumulators; 1 for mostly-unique batches (long
  // sequences) where extra splits are 8x wasted bounds probes.
  auto grad_unique = torch::empty(
      {m, dim}, grad_out.options().dtype(torch::kFloat32));
  if (m * dim == 0) return grad_unique;
  auto stream = current_stream();
  k_zero_f32<<<n_blocks((m * dim + 3) / 4), kBlock, 0, stream>>>(
      grad_unique.data_ptr<float>(), m * dim);
  const float* wptr =
      weights.defined() && weights.numel() ? weights.data_ptr<float>()
                                           : nullptr;
  const int32_t* mdp =
      m_dev.defined() && m_dev.numel() ? m_dev.data_ptr<int32_t>() : nullptr;
  if (grad_out.scalar_type() == torch::kBFloat16) {
    auto* gp = reinterpret_cast<const __hip_bfloat16*>(
        grad_out.data_ptr<at::BFloat16>());
    if (splits <= 1)
      launch_bwd_strided<__hip_bfloat16, 1>(
          gp, order, bounds, row_ids, wptr, row_coeff, m, mdp, batch,
          n_tables, dim, identity_rows, grad_unique, stream);
    else if (splits <= 8)
      launch_bwd_strided<__hip_bfloat16, 8>(
          gp, order, bounds, row_ids, wptr, row_coeff, m, mdp, batch,
          n_tables, dim, identity_rows, grad_unique, stream);
    else if (splits <= 16)
      launch_bwd_strided<__hip_bfloat16, 16, 64>(
          gp, order, bounds, row_ids, wptr, row_coeff, m, mdp, batch,
          n_tables, dim, identity_rows, grad_unique, stream);
    else if (splits <= 32)
      launch_bwd_strided<__hip_bfloat16, 32, 32>(
          gp, order, bounds, row_ids, wptr, row_coeff, m, mdp, batch,
          n_tables, dim, identity_rows, grad_unique, stream);
    else
      launch_bwd_strided<__hip_bfloat16, 64, 16>(
          gp, order, bounds, row_ids, wptr, row_coeff, m, mdp, batch,
          n_tables, dim, identity_rows, grad_unique, stream);
  } else {
    auto* gp = grad_out.data_ptr<float>();
    if (splits <= 1)
      launch_bwd_strided<float, 1>(
          gp, order, bounds, row_ids, wptr, row_coeff, m, mdp, batch,
          n_tables, dim, identity_rows, grad_unique, stream);
    else
      launch_bwd_strided<float, 8>(
          gp, order, bounds, row_ids, wptr, row_coeff, m, mdp, batch,
          n_tables, dim, identity_rows, grad_unique, stream);
  }
  return grad_unique;
}

// ---------------- sparse applies ----------------

void apply_sgd(torch::Tensor w, torch::Tensor slots, torch::Tensor grad,
               double lr) {
  int m = slots.numel();
  if (m == 0) return;
  int dim = w.size(1);
  auto stream = current_stream();
  k_apply_sgd<<<n_blocks((int64_t)m * dim), kBlock, 0, stream>>>(
      w.data_ptr<float>(), slots.data_ptr<int32_t>(), grad.data_ptr<float>(),
      m, dim, (float)lr);
}

void apply_adagrad(torch::Tensor w, torch::Tensor accum, torch::Tensor slots,
                   torch::Tensor grad, double lr, double epsilon) {
  int m = slots.numel();
  if (m == 0) return;
  int dim = w.size(1);
  auto stream = current_stream();
  k_apply_adagrad<<<n_blocks((int64_t)m * dim), kBlock, 0, stream>>>(
      w.data_ptr<float>(), accum.data_ptr<float>(), slots.data_ptr<int32_t>(),
      grad.data_ptr<float>(), m, dim, (float)lr, (float)epsilon);
}

void apply_adagrad_decay(torch::Tensor w, torch::Tensor accum,
                         torch::Tensor period_slab, torch::Tensor slots,
                         torch::Tensor grad, double lr, double epsilon,
                         double cur_period, double decay_rate,
                         double baseline) {
  int m = slots.numel();
  if (m == 0) return;
  int dim = w.size(1);
  auto stream = current_stream();
  k_apply_adagrad_decay<<<n_blocks((int64_t)m * dim), kBlock, 0, stream>>>(
      w.data_ptr<float>(), accum.data_ptr<float>(),
      period_slab.data_ptr<float>(), slots.data_ptr<int32_t>(),
      grad.data_ptr<float>(), m, dim, (float)lr, (float)epsilon,
      (float)cur_period, (float)decay_rate, (float)baseline);
  k_commit_period<<<n_blocks(m), kBlock, 0, stream>>>(
      period_slab.data_ptr<float>(), slots.data_ptr<int32_t>(), m,
      (float)cur_period);
}

void apply_adam(torch::Tensor w, torch::Tensor mom, torch::Tensor vel,
                torch::Tensor slots, torch::Tensor grad, double lr_t,
                double beta1, double beta2, double epsilon) {
  int m = slots.numel();
  if (m == 0) return;
  int dim = w.size(1);
  auto stream = current_stream();
  k_apply_adam<<<n_blocks((int64_t)m * dim), kBlock, 0, stream>>>(
      w.data_ptr<float>(), mom.data_ptr<float>(), vel.data_ptr<float>(),
      slots.data_ptr<int32_t>(), grad.data_ptr<float>(), m, dim, (float)lr_t,
      (float)beta1, (float)beta2, (float)epsilon);
}

void apply_adam_dev(torch::Tensor w, torch::Tensor mom, torch::Tensor vel,
                    torch::Tensor slots, torch::Tensor grad, double lr,
                    double beta1, double beta2, double epsilon,
                    torch::Tensor powers) {
  int m = slots.numel();
  if (m == 0) return;
  int dim = w.size(1);
  auto stream = current_stream();
  k_apply_adam_dev<<<n_blocks((int64_t)m * dim), kBlock, 0, stream>>>(
      w.data_ptr<float>(), mom.data_ptr<float>(), vel.data_ptr<float>(),
      slots.data_ptr<int32_t>(), grad.data_ptr<float>(), m, dim, (float)lr,
      (float)beta1, (float)beta2, (float)epsilon,
      powers.data_ptr<float>());
}

void update_powers(torch::Tensor powers, double beta1, double beta2) {
  auto stream = current_stream();
  k_update_powers<<<1, 64, 0, stream>>>(powers.data_ptr<float>(),
                                        (float)beta1, (float)beta2);
}

void apply_adamw(torch::Tensor w, torch::Tensor mom, torch::Tensor vel,
                 torch::Tensor slots, torch::Tensor grad, double lr_t,
                 double lr, double beta1, double beta2, double epsilon,
                 double weight_decay) {
  int m = slots.numel();
  if (m == 0) return;
  int dim = w.size(1);
  auto stream = current_stream();
  k_apply_adamw<<<n_blocks((int64_t)m * dim), kBlock, 0, stream>>>(
      w.data_ptr<float>(), mom.data_ptr<float>(), vel.data_ptr<float>(),
      slots.data_ptr<int32_t>(), grad.data_ptr<float>(), m, dim, (float)lr_t,
      (float)lr, (float)beta1, (float)beta2, (float)epsilon,
      (float)weight_decay);
}

void apply_rmsprop(torch::Tensor w, torch::Tensor vel, torch::Tensor slots,
                   torch::Tensor grad, double lr, double beta2,
                   double epsilon) {
  int m = slots.numel();
  if (m == 0) return;
  int dim = w.size(1);
  auto stream = current_stream();
  k_apply_rmsprop<<<n_blocks((int64_t)m * dim), kBlock, 0, stream>>>(
      w.data_ptr<float>(), vel.data_ptr<float>(), slots.data_ptr<int32_t>(),
      grad.data_ptr<float>(), m, dim, (float)lr, (float)beta2,
      (float)epsilon);
}

void apply_ftrl(torch::Tensor w, torch::Tensor n, torch::Tensor z,
                torch::Tensor slots, torch::Tensor grad, double lr, double l1,
                double l2, double lr_power, double l2_shrinkage) {
  int m = slots.numel();
  if (m == 0) return;
  int dim = w.size(1);
  auto stream = current_stream();
  k_apply_ftrl<<<n_blocks((int64_t)m * dim), kBlock, 0, stream>>>(
      w.data_ptr<float>(), n.data_ptr<float>(), z.data_ptr<float>(),
      slots.data_ptr<int32_t>(), grad.data_ptr<float>(), m, dim, (float)lr,
      (float)l1, (float)l2, (float)lr_power, (float)l2_shrinkage);
}

void register_dense(pybind11::module_& mod);      // dense_kernels.hip
void register_gru(pybind11::module_& mod);        // gru_kernels.hip
void register_attention(pybind11::module_& mod);  // attention_kernels.hip
void register_fp8(pybind11::module_& mod);        // fp8_kernels.hip

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  register_dense(mod);
  register_gru(mod);
  register_attention(mod);
  register_fp8(mod);
  mod.def("ht_lookup_insert", &ht_lookup_insert);
  mod.def("ht_dedup_a", &ht_dedup_a);
  mod.def("ht_dedup_b", &ht_dedup_b);
  mod.def("ht_dedup_b_padded", &ht_dedup_b_padded);
  mod.def("ht_dedup_a_dev", &ht_dedup_a_dev);
  mod.def("bump_epoch", &bump_epoch);
  mod.def("bump_epoch_only", &bump_epoch_only);
  mod.def("route_pad", &route_pad);
  mod.def("rows_to_padded", &rows_to_padded);
  mod.def("ht_dedup_c_pad", &ht_dedup_c_pad);
  mod.def("slots_gather_pad", &slots_gather_pad);
  mod.def("cnt_sum_pad", &cnt_sum_pad);
  mod.def("rows_segsum_pad", &rows_segsum_pad);
  mod.def("compose_i32", &compose_i32);
  mod.def("gather_host_rows", &gather_host_rows);
  mod.def("scatter_host_rows", &scatter_host_rows);
  mod.def("ht_dedup_c", &ht_dedup_c);
  mod.def("ht_dedup_c_idx", &ht_dedup_c_idx);
  mod.def("ht_dedup_c_idx_pad", &ht_dedup_c_idx_pad);
  mod.def("csr_order", &csr_order);
  mod.def("csr_scatter", &csr_scatter);
  mod.def("ht_insert_bulk", &ht_insert_bulk);
  mod.def("ht_lookup", &ht_lookup);
  mod.def("ht_export", &ht_export);
  mod.def("ev_gather", &ev_gather);
  mod.def("pooled_fwd", &pooled_fwd);
  mod.def("pooled_bwd", &pooled_bwd);
  mod.def("group_pooled_fwd", &group_pooled_fwd);
  mod.def("group_pooled_fwd_direct", &group_pooled_fwd_direct);
  mod.def("group_pooled_bwd_strided", &group_pooled_bwd_strided);
  mod.def("apply_sgd", &apply_sgd);
  mod.def("apply_adagrad", &apply_adagrad);
  mod.def("apply_adagrad_decay", &apply_adagrad_decay);
  mod.def("apply_adam", &apply_adam);
  mod.def("apply_adamw", &apply_adamw);
  mod.def("apply_adam_dev", &apply_adam_dev);
  mod.def("update_powers", &update_powers);
  mod.def("apply_rmsprop", &apply_rmsprop);
  mod.def("apply_ftrl", &apply_ftrl);
  mod.attr("EMPTY_KEY") = EMPTY_KEY;
  mod.attr("PAD_KEY") = INT64_MAX;
}
