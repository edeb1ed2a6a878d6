// CPU backend of the slab ops (ops.h) — used by the no-GPU test tier and
// by host-resident stores. Semantics identical to ops_hip.hip; concurrency
// is handled by the store's striped locks (the CPU path applies ops under
// the per-key stripe lock, so plain loads/stores suffice here).
#include <cstring>
#include "ops.h"

namespace adapm {

static inline const float* sel_base_c(const SlabBases& sb, int64_t& off) {
  if (off & SPILL_BIT) { off &= ~SPILL_BIT; return sb.host; }
  return sb.dev;
}
static inline float* sel_base(const SlabBases& sb, int64_t& off) {
  if (off & SPILL_BIT) { off &= ~SPILL_BIT; return sb.host; }
  return sb.dev;
}

void ops_gather_cpu(const SlabBases& sb, const OpsBatch& b, float* out) {
  for (int i = 0; i < b.n; ++i) {
    int64_t s = b.src_off[i];
    if (s < 0) continue;
    const float* slab = sel_base_c(sb, s);
    std::memcpy(out + b.dst_off[i], slab + s, sizeof(float) * b.lens[i]);
  }
}

void ops_scatter_cpu(const SlabBases& sb, const OpsBatch& b, const float* in, bool set) {
  for (int i = 0; i < b.n; ++i) {
    int64_t s = b.src_off[i];
    if (s < 0) continue;
    float* slab = sel_base(sb, s);
    int64_t d = b.dst_off[i];
    int32_t len = b.lens[i];
    if (set) {
      std::memcpy(slab + s, in + d, sizeof(float) * len);
    } else {
      for (int e = 0; e < len; ++e) slab[s + e] += in[d + e];
    }
  }
}

void ops_extract_cpu(const SlabBases& sb, const OpsBatch& b, const int64_t* sync_off, float* out) {
  for (int i = 0; i < b.n; ++i) {
    int64_t v = b.src_off[i];
    if (v < 0) continue;
    int64_t o = b.dst_off[i], sy = sync_off[i];
    float* vb = sel_base(sb, v);
    float* syb = sel_base(sb, sy);
    int32_t len = b.lens[i];
    for (int e = 0; e < len; ++e) {
      float cur = vb[v + e];
      out[o + e] = cur - syb[sy + e];
      syb[sy + e] = cur;
    }
  }
}

void ops_refresh_cpu(const SlabBases& sb, const OpsBatch& b, const int64_t* sync_off, const float* state_in) {
  for (int i = 0; i < b.n; ++i) {
    int64_t v = b.src_off[i];
    if (v < 0) continue;
    int64_t o = b.dst_off[i], sy = sync_off[i];
    float* vb = sel_base(sb, v);
    float* syb = sel_base(sb, sy);
    int32_t len = b.lens[i];
    for (int e = 0; e < len; ++e) {
      float s = state_in[o + e];
      vb[v + e] += s - syb[sy + e];
      syb[sy + e] = s;
    }
  }
}

void ops_gather_keys_cpu(const SlabBases& sb, const KeyBatch& b, float* out) {
  for (int i = 0; i < b.n; ++i) {
    int64_t k = b.keys[i];
    if ((int)(k % b.world) != b.rank) continue;
    std::memcpy(out + (int64_t)i * b.len, sb.dev + (k / b.world) * (int64_t)b.plen,
                sizeof(float) * b.len);
  }
}

void ops_scatter_keys_cpu(const SlabBases& sb, const KeyBatch& b, const float* in, bool set) {
  for (int i = 0; i < b.n; ++i) {
    int64_t k = b.keys[i];
    if ((int)(k % b.world) != b.rank) continue;
    float* s = sb.dev + (k / b.world) * (int64_t)b.plen;
    const float* d = in + (int64_t)i * b.len;
    if (set) {
      std::memcpy(s, d, sizeof(float) * b.len);
    } else {
      for (int e = 0; e < b.len; ++e) s[e] += d[e];
    }
  }
}

void ops_zero_cpu(const SlabBases& sb, const OpsBatch& b) {
  for (int i = 0; i < b.n; ++i) {
    int64_t d = b.dst_off[i];
    if (d < 0) continue;
    float* slab = sel_base(sb, d);
    std::memset(slab + d, 0, sizeof(float) * b.lens[i]);
  }
}

}  // namespace adapm

namespace adapm {
void ops_delta_sqnorm_cpu(const SlabBases& slab, const OpsBatch& b, const int64_t* sync_off,
                          float* out) {
  for (int i = 0; i < b.n; ++i) {
    int64_t v = b.src_off[i], sy = sync_off[i];
    const float* vb = sel_base_c(slab, v);
    const float* sb = sel_base_c(slab, sy);
    float acc = 0.f;
    for (int e = 0; e < b.lens[i]; ++e) {
      float d = vb[v + e] - sb[sy + e];
      acc += d * d;
    }
    out[b.dst_off[i]] = acc;
  }
}
}  // namespace adapm
