// Batched slab ops — the hot loops of the parameter store.
//
// These replace the reference's scalar host loops (mergeValue / pull-copy /
// delta-extract / replica-refresh, coloc_kv_server_handle.h:404-415, 464,
// 630-648, 789-809) with batched kernels over the HBM slab. Each op takes
// explicit per-key slab offsets computed by the host metadata pass, so the
// device needs no key metadata at all.
//
// Two backends with identical semantics:
//  - HIP/gfx950 (ops_hip.hip) — one workgroup per key, float4-vectorized,
//    atomicAdd for concurrent-push safety.
//  - CPU (ops_cpu.cpp) — plain loops, used by the no-GPU test tier.
#pragma once
#include <cstdint>

namespace adapm {

// offsets with bit 62 set live in the host-spill arena (pinned host
// memory, device-visible zero-copy): kernels decode the bit and pick the
// base pointer. See Slab host spill in core.cpp.
constexpr int64_t SPILL_BIT = 1LL << 62;

struct SlabBases {
  float* dev;   // HBM slab
  float* host;  // device-visible pointer to the pinned host arena (may be null)
};

struct OpsBatch {
  const int64_t* src_off;  // per-key slab offset (floats), or -1 to skip
  const int64_t* dst_off;  // per-key offset into the out/in buffer (floats)
  const int32_t* lens;     // per-key value length (floats)
  int n;                   // number of keys in the batch
};

// out[dst_off[i] : +len] = slab[src_off[i] : +len]
void ops_gather_gpu(const SlabBases& slab, const OpsBatch& b, float* out, void* stream);
void ops_gather_cpu(const SlabBases& slab, const OpsBatch& b, float* out);

// slab[dst(src)_off[i]] += in[...]   (atomic on GPU)  — or assign when set=true
void ops_scatter_gpu(const SlabBases& slab, const OpsBatch& b, const float* in, bool set, void* stream);
void ops_scatter_cpu(const SlabBases& slab, const OpsBatch& b, const float* in, bool set);
// merge (+=) with plain load/store instead of atomics. ONLY safe when no
// two batch entries share a destination slot. Exists for host-spilled
// rows: an atomicAdd to pinned host memory is a non-posted PCIe
// round-trip per dword (measured 144x slower than the zero-copy read),
// while a read+add+posted-write streams at PCIe bandwidth.
void ops_scatter_rmw_gpu(const SlabBases& slab, const OpsBatch& b, const float* in, void* stream);
// per-key squared L2 norm of the pending delta (val - sync_state):
// out[dst_off[i]] = ||slab[src_off[i]..] - slab[sync_off[i]..]||^2.
// Drives the reference's --sys.sync.threshold ("send only deltas whose
// norm clears the threshold"; sync_manager.h:601-662) without reading
// the values back to the host.
void ops_delta_sqnorm_gpu(const SlabBases& slab, const OpsBatch& b, const int64_t* sync_off,
                          float* out, void* stream);
void ops_delta_sqnorm_cpu(const SlabBases& slab, const OpsBatch& b, const int64_t* sync_off,
                          float* out);

// replica delta extraction (src_off = val offsets, dst_off = out buffer
// offsets, aux_off = sync_state offsets):
//   v = val[e]; out[e] = v - sync[e]; sync[e] = v
// single-read-per-element so a concurrent atomic push is never lost (it
// stays in val and is extracted next round).
void ops_extract_gpu(const SlabBases& slab, const OpsBatch& b, const int64_t* sync_off, float* out, void* stream);
void ops_extract_cpu(const SlabBases& slab, const OpsBatch& b, const int64_t* sync_off, float* out);

// replica refresh apply (delta form so concurrent pushes are preserved):
//   s = state_in[e]; atomicAdd(&val[e], s - sync[e]); sync[e] = s
void ops_refresh_gpu(const SlabBases& slab, const OpsBatch& b, const int64_t* sync_off, const float* state_in, void* stream);
void ops_refresh_cpu(const SlabBases& slab, const OpsBatch& b, const int64_t* sync_off, const float* state_in);

// slab[dst_off[i] : +len] = 0
void ops_zero_gpu(const SlabBases& slab, const OpsBatch& b, void* stream);
void ops_zero_cpu(const SlabBases& slab, const OpsBatch& b);

// Identity-layout direct ops: key k (owned: k % world == rank) lives at
// slab offset (k / world) * plen; out/in row i sits at i * len. Keys not
// owned here are skipped (the host enqueues them as remote ops). This is
// the zero-host-work fast path: keys go H2D and the kernel derives every
// offset itself.
struct KeyBatch {
  const int64_t* keys;  // device (gpu) / host (cpu)
  int n;
  int32_t len;    // uniform value length (floats)
  int32_t plen;   // padded slot length
  int world, rank;
};
void ops_gather_keys_gpu(const SlabBases& slab, const KeyBatch& b, float* out, void* stream);
void ops_gather_keys_cpu(const SlabBases& slab, const KeyBatch& b, float* out);
void ops_scatter_keys_gpu(const SlabBases& slab, const KeyBatch& b, const float* in, bool set, void* stream);
void ops_scatter_keys_cpu(const SlabBases& slab, const KeyBatch& b, const float* in, bool set);

// Sorted scatter-add (identity layout): keys pre-sorted on device with the
// permutation into `in` rows. Consecutive duplicate keys are pre-summed in
// registers, so hot keys (Zipf pushes) cost one atomic per 16-row chunk
// instead of one per row — removes atomic contention.
void ops_scatter_sorted_gpu(const SlabBases& slab, const int64_t* sorted_keys,
                            const int64_t* perm, int n, int32_t len, int32_t plen, int world,
                            int rank, const float* in, void* stream);

bool hip_available();

}  // namespace adapm
