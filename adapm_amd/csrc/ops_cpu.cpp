// CPU backend of the slab ops (ops.h) — used by the no-GPU test tier and
// by host-resident stores. Semantics identical to ops_hip.hip; concurrency
// is handled by the store's striped locks (the CPU path applies ops under
// the per-key stripe lock, so plain loads/stores suffice here).
#include <cstring>
#include "ops.h"

namespace adapm {

void ops_gather_cpu(const float* slab, const OpsBatch& b, float* out) {
  for (int i = 0; i < b.n; ++i) {
    int64_t s = b.src_off[i];
    if (s < 0) continue;
    std::memcpy(out + b.dst_off[i], slab + s, sizeof(float) * b.lens[i]);
  }
}

void ops_scatter_cpu(float* slab, const OpsBatch& b, const float* in, bool set) {
  for (int i = 0; i < b.n; ++i) {
    int64_t s = b.src_off[i];
    if (s < 0) continue;
    int64_t d = b.dst_off[i];
    int32_t len = b.lens[i];
    if (set) {
      std::memcpy(slab + s, in + d, sizeof(float) * len);
    } else {
      for (int e = 0; e < len; ++e) slab[s + e] += in[d + e];
    }
  }
}

void ops_extract_cpu(float* slab, const OpsBatch& b, const int64_t* sync_off, float* out) {
  for (int i = 0; i < b.n; ++i) {
    int64_t v = b.src_off[i];
    if (v < 0) continue;
    int64_t o = b.dst_off[i], sy = sync_off[i];
    int32_t len = b.lens[i];
    for (int e = 0; e < len; ++e) {
      float cur = slab[v + e];
      out[o + e] = cur - slab[sy + e];
      slab[sy + e] = cur;
    }
  }
}

void ops_refresh_cpu(float* slab, const OpsBatch& b, const int64_t* sync_off, const float* state_in) {
  for (int i = 0; i < b.n; ++i) {
    int64_t v = b.src_off[i];
    if (v < 0) continue;
    int64_t o = b.dst_off[i], sy = sync_off[i];
    int32_t len = b.lens[i];
    for (int e = 0; e < len; ++e) {
      float s = state_in[o + e];
      slab[v + e] += s - slab[sy + e];
      slab[sy + e] = s;
    }
  }
}

void ops_gather_keys_cpu(const float* slab, const KeyBatch& b, float* out) {
  for (int i = 0; i < b.n; ++i) {
    int64_t k = b.keys[i];
    if ((int)(k % b.world) != b.rank) continue;
    std::memcpy(out + (int64_t)i * b.len, slab + (k / b.world) * (int64_t)b.plen,
                sizeof(float) * b.len);
  }
}

void ops_scatter_keys_cpu(float* slab, const KeyBatch& b, const float* in, bool set) {
  for (int i = 0; i < b.n; ++i) {
    int64_t k = b.keys[i];
    if ((int)(k % b.world) != b.rank) continue;
    float* s = slab + (k / b.world) * (int64_t)b.plen;
    const float* d = in + (int64_t)i * b.len;
    if (set) {
      std::memcpy(s, d, sizeof(float) * b.len);
    } else {
      for (int e = 0; e < b.len; ++e) s[e] += d[e];
    }
  }
}

void ops_zero_cpu(float* slab, const OpsBatch& b) {
  for (int i = 0; i < b.n; ++i) {
    int64_t d = b.dst_off[i];
    if (d < 0) continue;
    std::memset(slab + d, 0, sizeof(float) * b.lens[i]);
  }
}

}  // namespace adapm
