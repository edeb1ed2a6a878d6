// HIP/gfx950 slab ops — see ops.h for semantics.
//
// Design (CDNA4): one 256-thread workgroup per key, grid-stride over keys
// so batches of any size fill the 256 CUs; float4 when the slot is
// 16B-aligned (allocator pads every slot to a multiple of 4 floats, so the
// vector path is the common case); scalar tail otherwise. All slab writes
// that can race with concurrent worker pushes are atomicAdd or follow the
// single-read discipline documented in ops.h.
#include <hip/hip_runtime.h>
#include "ops.h"

namespace adapm {

#define THREADS 256

__device__ inline float* sel_base(float* dev, float* host, int64_t& off) {
  if (off & SPILL_BIT) { off &= ~SPILL_BIT; return host; }
  return dev;
}

__device__ inline bool vec_ok(int64_t a, int64_t b, int32_t len) {
  return ((a | b) & 3) == 0 && (len & 3) == 0;
}

__global__ void k_gather(float* __restrict__ sdev, float* __restrict__ shost,
                         const int64_t* __restrict__ src_off,
                         const int64_t* __restrict__ dst_off, const int32_t* __restrict__ lens,
                         int n, float* __restrict__ out) {
  for (int i = blockIdx.x; i < n; i += gridDim.x) {
    int64_t s = src_off[i];
    if (s < 0) continue;
    const float* slab = sel_base(sdev, shost, s);
    int64_t d = dst_off[i];
    int32_t len = lens[i];
    if (vec_ok(s, d, len)) {
      const float4* sp = reinterpret_cast<const float4*>(slab + s);
      float4* dp = reinterpret_cast<float4*>(out + d);
      for (int e = threadIdx.x; e < (len >> 2); e += THREADS) dp[e] = sp[e];
    } else {
      for (int e = threadIdx.x; e < len; e += THREADS) out[d + e] = slab[s + e];
    }
  }
}

__global__ void k_scatter_add(float* __restrict__ sdev, float* __restrict__ shost,
                              const int64_t* __restrict__ src_off,
                              const int64_t* __restrict__ dst_off, const int32_t* __restrict__ lens,
                              int n, const float* __restrict__ in) {
  for (int i = blockIdx.x; i < n; i += gridDim.x) {
    int64_t s = src_off[i];
    if (s < 0) continue;
    float* slab = sel_base(sdev, shost, s);
    int64_t d = dst_off[i];
    int32_t len = lens[i];
    for (int e = threadIdx.x; e < len; e += THREADS) atomicAdd(&slab[s + e], in[d + e]);
  }
}

// non-atomic merge for unique destination slots (host-spill fast path;
// see ops.h) — vectorized read+add+posted-write
__global__ void k_scatter_rmw(float* __restrict__ sdev, float* __restrict__ shost,
                              const int64_t* __restrict__ src_off,
                              const int64_t* __restrict__ dst_off, const int32_t* __restrict__ lens,
                              int n, const float* __restrict__ in) {
  for (int i = blockIdx.x; i < n; i += gridDim.x) {
    int64_t s = src_off[i];
    if (s < 0) continue;
    float* slab = sel_base(sdev, shost, s);
    int64_t d = dst_off[i];
    int32_t len = lens[i];
    if (vec_ok(s, d, len)) {
      float4* sp = reinterpret_cast<float4*>(slab + s);
      const float4* dp = reinterpret_cast<const float4*>(in + d);
      for (int e = threadIdx.x; e < (len >> 2); e += THREADS) {
        float4 v = sp[e], w = dp[e];
        v.x += w.x; v.y += w.y; v.z += w.z; v.w += w.w;
        sp[e] = v;
      }
    } else {
      for (int e = threadIdx.x; e < len; e += THREADS) slab[s + e] += in[d + e];
    }
  }
}

__global__ void k_scatter_set(float* __restrict__ sdev, float* __restrict__ shost,
                              const int64_t* __restrict__ src_off,
                              const int64_t* __restrict__ dst_off, const int32_t* __restrict__ lens,
                              int n, const float* __restrict__ in) {
  for (int i = blockIdx.x; i < n; i += gridDim.x) {
    int64_t s = src_off[i];
    if (s < 0) continue;
    float* slab = sel_base(sdev, shost, s);
    int64_t d = dst_off[i];
    int32_t len = lens[i];
    if (vec_ok(s, d, len)) {
      const float4* ip = reinterpret_cast<const float4*>(in + d);
      float4* sp = reinterpret_cast<float4*>(slab + s);
      for (int e = threadIdx.x; e < (len >> 2); e += THREADS) sp[e] = ip[e];
    } else {
      for (int e = threadIdx.x; e < len; e += THREADS) slab[s + e] = in[d + e];
    }
  }
}

__global__ void k_extract(float* __restrict__ sdev, float* __restrict__ shost,
                          const int64_t* __restrict__ val_off,
                          const int64_t* __restrict__ out_off, const int32_t* __restrict__ lens,
                          int n, const int64_t* __restrict__ sync_off, float* __restrict__ out) {
  for (int i = blockIdx.x; i < n; i += gridDim.x) {
    int64_t v = val_off[i];
    if (v < 0) continue;
    int64_t o = out_off[i], sy = sync_off[i];
    float* vb = sel_base(sdev, shost, v);
    float* sb = sel_base(sdev, shost, sy);
    int32_t len = lens[i];
    for (int e = threadIdx.x; e < len; e += THREADS) {
      float cur = vb[v + e];        // single read: a concurrent atomicAdd
      out[o + e] = cur - sb[sy + e];  // after this read stays in val and is
      sb[sy + e] = cur;             // captured by the next round's extract
    }
  }
}

__global__ void k_refresh(float* __restrict__ sdev, float* __restrict__ shost,
                          const int64_t* __restrict__ val_off,
                          const int64_t* __restrict__ in_off, const int32_t* __restrict__ lens,
                          int n, const int64_t* __restrict__ sync_off,
                          const float* __restrict__ state_in) {
  for (int i = blockIdx.x; i < n; i += gridDim.x) {
    int64_t v = val_off[i];
    if (v < 0) continue;
    int64_t o = in_off[i], sy = sync_off[i];
    float* vb = sel_base(sdev, shost, v);
    float* sb = sel_base(sdev, shost, sy);
    int32_t len = lens[i];
    for (int e = threadIdx.x; e < len; e += THREADS) {
      float s = state_in[o + e];
      atomicAdd(&vb[v + e], s - sb[sy + e]);  // delta form: preserves
      sb[sy + e] = s;                         // concurrent pushes
    }
  }
}

__global__ void k_zero(float* __restrict__ sdev, float* __restrict__ shost,
                       const int64_t* __restrict__ dst_off,
                       const int32_t* __restrict__ lens, int n) {
  for (int i = blockIdx.x; i < n; i += gridDim.x) {
    int64_t d = dst_off[i];
    if (d < 0) continue;
    float* slab = sel_base(sdev, shost, d);
    int32_t len = lens[i];
    for (int e = threadIdx.x; e < len; e += THREADS) slab[d + e] = 0.f;
  }
}

__global__ void k_gather_keys(const float* __restrict__ slab, const int64_t* __restrict__ keys,
                              int n, int32_t len, int32_t plen, int world, int rank,
                              float* __restrict__ out) {
  for (int i = blockIdx.x; i < n; i += gridDim.x) {
    int64_t k = keys[i];
    if ((int)(k % world) != rank) continue;
    const float4* sp = reinterpret_cast<const float4*>(slab + (k / world) * (int64_t)plen);
    float4* dp = reinterpret_cast<float4*>(out + (int64_t)i * len);
    if ((len & 3) == 0) {
      for (int e = threadIdx.x; e < (len >> 2); e += THREADS) dp[e] = sp[e];
    } else {
      const float* s = reinterpret_cast<const float*>(sp);
      float* d = reinterpret_cast<float*>(dp);
      for (int e = threadIdx.x; e < len; e += THREADS) d[e] = s[e];
    }
  }
}

__global__ void k_scatter_add_keys(float* __restrict__ slab, const int64_t* __restrict__ keys,
                                   int n, int32_t len, int32_t plen, int world, int rank,
                                   const float* __restrict__ in) {
  for (int i = blockIdx.x; i < n; i += gridDim.x) {
    int64_t k = keys[i];
    if ((int)(k % world) != rank) continue;
    float* s = slab + (k / world) * (int64_t)plen;
    const float* d = in + (int64_t)i * len;
    for (int e = threadIdx.x; e < len; e += THREADS) atomicAdd(&s[e], d[e]);
  }
}

__global__ void k_scatter_set_keys(float* __restrict__ slab, const int64_t* __restrict__ keys,
                                   int n, int32_t len, int32_t plen, int world, int rank,
                                   const float* __restrict__ in) {
  for (int i = blockIdx.x; i < n; i += gridDim.x) {
    int64_t k = keys[i];
    if ((int)(k % world) != rank) continue;
    float* s = slab + (k / world) * (int64_t)plen;
    const float* d = in + (int64_t)i * len;
    if ((len & 3) == 0) {
      float4* sp = reinterpret_cast<float4*>(s);
      const float4* dp = reinterpret_cast<const float4*>(d);
      for (int e = threadIdx.x; e < (len >> 2); e += THREADS) sp[e] = dp[e];
    } else {
      for (int e = threadIdx.x; e < len; e += THREADS) s[e] = d[e];
    }
  }
}

static inline int grid_for(int n) {
  // >=2048 workgroups fills 256 CUs at 8 blocks/CU; grid-stride covers the rest
  int g = n < 1 ? 1 : n;
  return g > 16384 ? 16384 : g;
}

void ops_gather_gpu(const SlabBases& slab, const OpsBatch& b, float* out, void* stream) {
  if (b.n == 0) return;
  hipLaunchKernelGGL(k_gather, dim3(grid_for(b.n)), dim3(THREADS), 0, (hipStream_t)stream,
                     slab.dev, slab.host, b.src_off, b.dst_off, b.lens, b.n, out);
}
void ops_scatter_gpu(const SlabBases& slab, const OpsBatch& b, const float* in, bool set, void* stream) {
  if (b.n == 0) return;
  if (set)
    hipLaunchKernelGGL(k_scatter_set, dim3(grid_for(b.n)), dim3(THREADS), 0, (hipStream_t)stream,
                       slab.dev, slab.host, b.src_off, b.dst_off, b.lens, b.n, in);
  else
    hipLaunchKernelGGL(k_scatter_add, dim3(grid_for(b.n)), dim3(THREADS), 0, (hipStream_t)stream,
                       slab.dev, slab.host, b.src_off, b.dst_off, b.lens, b.n, in);
}
void ops_scatter_rmw_gpu(const SlabBases& slab, const OpsBatch& b, const float* in, void* stream) {
  if (b.n == 0) return;
  hipLaunchKernelGGL(k_scatter_rmw, dim3(grid_for(b.n)), dim3(THREADS), 0, (hipStream_t)stream,
                     slab.dev, slab.host, b.src_off, b.dst_off, b.lens, b.n, in);
}
__global__ void k_delta_sqnorm(float* __restrict__ sdev, float* __restrict__ shost,
                               const int64_t* __restrict__ src_off,
                               const int64_t* __restrict__ dst_off,
                               const int32_t* __restrict__ lens, int n,
                               const int64_t* __restrict__ sync_off, float* __restrict__ out) {
  __shared__ float red[THREADS / 64];
  for (int i = blockIdx.x; i < n; i += gridDim.x) {
    int64_t v = src_off[i], sy = sync_off[i];
    const float* vb = sel_base(sdev, shost, v);
    const float* sb = sel_base(sdev, shost, sy);
    int32_t len = lens[i];
    float acc = 0.f;
    for (int e = threadIdx.x; e < len; e += THREADS) {
      float d = vb[v + e] - sb[sy + e];
      acc += d * d;
    }
    // wave64 reduce then cross-wave via LDS
    for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = acc;
    __syncthreads();
    if (threadIdx.x == 0) {
      float t = 0.f;
      for (int w = 0; w < THREADS / 64; ++w) t += red[w];
      out[dst_off[i]] = t;
    }
    __syncthreads();
  }
}

void ops_delta_sqnorm_gpu(const SlabBases& slab, const OpsBatch& b, const int64_t* sync_off,
                          float* out, void* stream) {
  if (b.n == 0) return;
  hipLaunchKernelGGL(k_delta_sqnorm, dim3(grid_for(b.n)), dim3(THREADS), 0, (hipStream_t)stream,
                     slab.dev, slab.host, b.src_off, b.dst_off, b.lens, b.n, sync_off, out);
}

void ops_extract_gpu(const SlabBases& slab, const OpsBatch& b, const int64_t* sync_off, float* out, void* stream) {
  if (b.n == 0) return;
  hipLaunchKernelGGL(k_extract, dim3(grid_for(b.n)), dim3(THREADS), 0, (hipStream_t)stream,
                     slab.dev, slab.host, b.src_off, b.dst_off, b.lens, b.n, sync_off, out);
}
void ops_refresh_gpu(const SlabBases& slab, const OpsBatch& b, const int64_t* sync_off, const float* state_in, void* stream) {
  if (b.n == 0) return;
  hipLaunchKernelGGL(k_refresh, dim3(grid_for(b.n)), dim3(THREADS), 0, (hipStream_t)stream,
                     slab.dev, slab.host, b.src_off, b.dst_off, b.lens, b.n, sync_off, state_in);
}
void ops_zero_gpu(const SlabBases& slab, const OpsBatch& b, void* stream) {
  if (b.n == 0) return;
  hipLaunchKernelGGL(k_zero, dim3(grid_for(b.n)), dim3(THREADS), 0, (hipStream_t)stream,
                     slab.dev, slab.host, b.dst_off, b.lens, b.n);
}

void ops_gather_keys_gpu(const SlabBases& slab, const KeyBatch& b, float* out, void* stream) {
  if (b.n == 0) return;  // identity layout never spills
  hipLaunchKernelGGL(k_gather_keys, dim3(grid_for(b.n)), dim3(THREADS), 0, (hipStream_t)stream,
                     slab.dev, b.keys, b.n, b.len, b.plen, b.world, b.rank, out);
}
void ops_scatter_keys_gpu(const SlabBases& slab, const KeyBatch& b, const float* in, bool set, void* stream) {
  if (b.n == 0) return;
  if (set)
    hipLaunchKernelGGL(k_scatter_set_keys, dim3(grid_for(b.n)), dim3(THREADS), 0,
                       (hipStream_t)stream, slab.dev, b.keys, b.n, b.len, b.plen, b.world, b.rank, in);
  else
    hipLaunchKernelGGL(k_scatter_add_keys, dim3(grid_for(b.n)), dim3(THREADS), 0,
                       (hipStream_t)stream, slab.dev, b.keys, b.n, b.len, b.plen, b.world, b.rank, in);
}

bool hip_available() {
  int n = 0;
  return hipGetDeviceCount(&n) == hipSuccess && n > 0;
}

}  // namespace adapm

namespace adapm {

#define SCHUNK 16  // sorted rows per workgroup

__global__ void k_scatter_add_sorted(float* __restrict__ slab, const int64_t* __restrict__ skeys,
                                     const int64_t* __restrict__ perm, int n, int32_t len,
                                     int32_t plen, int world, int rank,
                                     const float* __restrict__ in) {
  __shared__ int64_t offs[SCHUNK];
  __shared__ int32_t rows[SCHUNK];
  for (int c0 = blockIdx.x * SCHUNK; c0 < n; c0 += gridDim.x * SCHUNK) {
    int cn = min(SCHUNK, n - c0);
    if (threadIdx.x < (unsigned)cn) {
      int64_t k = skeys[c0 + threadIdx.x];
      offs[threadIdx.x] =
          ((int)(k % world) == rank) ? (k / world) * (int64_t)plen : (int64_t)-1;
      rows[threadIdx.x] = (int32_t)perm[c0 + threadIdx.x];
    }
    __syncthreads();
    for (int e = threadIdx.x; e < len; e += THREADS) {
      float acc = 0.f;
      int64_t cur = -1;
      for (int j = 0; j < cn; ++j) {
        int64_t o = offs[j];
        if (o < 0) continue;
        float v = in[(int64_t)rows[j] * len + e];
        if (o != cur) {
          if (cur >= 0) atomicAdd(&slab[cur + e], acc);
          cur = o;
          acc = 0.f;
        }
        acc += v;
      }
      if (cur >= 0) atomicAdd(&slab[cur + e], acc);
    }
    __syncthreads();
  }
}

void ops_scatter_sorted_gpu(const SlabBases& slab, const int64_t* sorted_keys,
                            const int64_t* perm, int n, int32_t len, int32_t plen, int world,
                            int rank, const float* in, void* stream) {
  if (n == 0) return;
  int blocks = (int)std::min<int64_t>(((int64_t)n + SCHUNK - 1) / SCHUNK, 16384);
  hipLaunchKernelGGL(k_scatter_add_sorted, dim3(blocks), dim3(THREADS), 0, (hipStream_t)stream,
                     slab.dev, sorted_keys, perm, n, len, plen, world, rank, in);
}

}  // namespace adapm
