// HIP/gfx950 app kernels — see kernels.h. One workgroup (256 threads =
// 4 waves) per training sample; rows are 2D floats = [emb(D) | accum(D)].
// These kernels are HBM-bound (each pulled row is read once, each delta
// row written once); the arithmetic per element is small, so the design
// goal is coalesced float4 row traffic and cheap block reductions
// (wave __shfl_xor + one LDS hop across the 4 waves).
#include <hip/hip_runtime.h>
#include <algorithm>

#include "kernels.h"

namespace adapm {

#define KT 256  // threads per workgroup

__device__ inline float block_reduce_sum(float v, float* lds) {
  // wave-level reduce (64 lanes)
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  int wave = threadIdx.x >> 6;
  int lane = threadIdx.x & 63;
  if (lane == 0) lds[wave] = v;
  __syncthreads();
  float r = (threadIdx.x < KT / 64) ? lds[threadIdx.x] : 0.f;
  if (threadIdx.x < 64) {
    for (int off = 2; off > 0; off >>= 1) r += __shfl_xor(r, off, 64);
  }
  if (threadIdx.x == 0) lds[0] = r;
  __syncthreads();
  float out = lds[0];
  __syncthreads();
  return out;
}

__device__ inline float sigmoidf(float x) { return 1.f / (1.f + __expf(-x)); }
__device__ inline float softplusf(float x) {
  // log(1+exp(x)), stable
  return x > 20.f ? x : __logf(1.f + __expf(x));
}

// --------------------------------------------------------------- ComplEx

// Per workgroup: one positive triple + its N negatives (o-side corruption).
// Each thread owns complex positions k = tid + i*KT, i < KPT (compile-time
// KPT so the per-thread arrays stay in registers — runtime-indexed arrays
// spill to scratch on gfx950). D = 512 -> dc = 256 -> KPT = 1.
template <int KPT>
__global__ void k_kge_step(const float* __restrict__ s, const float* __restrict__ r,
                           const float* __restrict__ o, const float* __restrict__ neg,
                           float* __restrict__ ds, float* __restrict__ dr,
                           float* __restrict__ do_, float* __restrict__ dneg,
                           float* __restrict__ loss, int B, int N, int D, float lr, float eps) {
  __shared__ float lds[KT / 64];
  const int dc = D >> 1;
  const int row = D << 1;  // floats per pulled row
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    const float* sb = s + (int64_t)b * row;
    const float* rb = r + (int64_t)b * row;

    float s_re[KPT], s_im[KPT], r_re[KPT], r_im[KPT];
    float a_sre[KPT], a_sim[KPT], a_rre[KPT], a_rim[KPT];
    float u_re[KPT], u_im[KPT];
#pragma unroll
    for (int i = 0; i < KPT; ++i) {
      int k = threadIdx.x + i * KT;
      a_sre[i] = a_sim[i] = a_rre[i] = a_rim[i] = 0.f;
      if (k < dc) {
        s_re[i] = sb[k];
        s_im[i] = sb[dc + k];
        r_re[i] = rb[k];
        r_im[i] = rb[dc + k];
        // per-object o-gradient direction (independent of o)
        u_re[i] = s_re[i] * r_re[i] - s_im[i] * r_im[i];
        u_im[i] = s_im[i] * r_re[i] + s_re[i] * r_im[i];
      } else {
        s_re[i] = s_im[i] = r_re[i] = r_im[i] = u_re[i] = u_im[i] = 0.f;
      }
    }

    float lsum = 0.f;
    for (int j = 0; j <= N; ++j) {
      const float* ob = (j == 0) ? o + (int64_t)b * row
                                 : neg + ((int64_t)b * N + (j - 1)) * row;
      float y = (j == 0) ? 1.f : -1.f;
      float part = 0.f;
      float o_re[KPT], o_im[KPT];
#pragma unroll
      for (int i = 0; i < KPT; ++i) {
        int k = threadIdx.x + i * KT;
        o_re[i] = k < dc ? ob[k] : 0.f;
        o_im[i] = k < dc ? ob[dc + k] : 0.f;
        part += u_re[i] * o_re[i] + u_im[i] * o_im[i];
      }
      float psi = block_reduce_sum(part, lds);
      float c = -y * sigmoidf(-y * psi);  // dL/dpsi, L = softplus(-y*psi)
      if (threadIdx.x == 0) lsum += softplusf(-y * psi);

      float* dob = (j == 0) ? do_ + (int64_t)b * row
                            : dneg + ((int64_t)b * N + (j - 1)) * row;
#pragma unroll
      for (int i = 0; i < KPT; ++i) {
        int k = threadIdx.x + i * KT;
        if (k >= dc) continue;
        // accumulate s/r grads
        a_sre[i] += c * (r_re[i] * o_re[i] + r_im[i] * o_im[i]);
        a_sim[i] += c * (r_re[i] * o_im[i] - r_im[i] * o_re[i]);
        a_rre[i] += c * (s_re[i] * o_re[i] + s_im[i] * o_im[i]);
        a_rim[i] += c * (s_re[i] * o_im[i] - s_im[i] * o_re[i]);
        // o grad + fused AdaGrad (G in the row's second half)
        float g_re = c * u_re[i];
        float g_im = c * u_im[i];
        float G_re = ob[D + k] + g_re * g_re;
        float G_im = ob[D + dc + k] + g_im * g_im;
        dob[k] = -lr * g_re * __frsqrt_rn(G_re + eps);
        dob[dc + k] = -lr * g_im * __frsqrt_rn(G_im + eps);
        dob[D + k] = g_re * g_re;
        dob[D + dc + k] = g_im * g_im;
      }
    }
    // write s/r deltas with fused AdaGrad
    float* dsb = ds + (int64_t)b * row;
    float* drb = dr + (int64_t)b * row;
#pragma unroll
    for (int i = 0; i < KPT; ++i) {
      int k = threadIdx.x + i * KT;
      if (k >= dc) continue;
      float Gsr = sb[D + k] + a_sre[i] * a_sre[i];
      float Gsi = sb[D + dc + k] + a_sim[i] * a_sim[i];
      dsb[k] = -lr * a_sre[i] * __frsqrt_rn(Gsr + eps);
      dsb[dc + k] = -lr * a_sim[i] * __frsqrt_rn(Gsi + eps);
      dsb[D + k] = a_sre[i] * a_sre[i];
      dsb[D + dc + k] = a_sim[i] * a_sim[i];
      float Grr = rb[D + k] + a_rre[i] * a_rre[i];
      float Gri = rb[D + dc + k] + a_rim[i] * a_rim[i];
      drb[k] = -lr * a_rre[i] * __frsqrt_rn(Grr + eps);
      drb[dc + k] = -lr * a_rim[i] * __frsqrt_rn(Gri + eps);
      drb[D + k] = a_rre[i] * a_rre[i];
      drb[D + dc + k] = a_rim[i] * a_rim[i];
    }
    if (threadIdx.x == 0) loss[b] = lsum;
  }
}

__global__ void k_kge_score(const float* __restrict__ s, const float* __restrict__ r,
                            const float* __restrict__ cand, float* __restrict__ scores, int B,
                            int E, int D) {
  __shared__ float lds[KT / 64];
  const int dc = D >> 1;
  const int row = D << 1;
  for (int be = blockIdx.x; be < B * E; be += gridDim.x) {
    int b = be / E, e = be % E;
    const float* sb = s + (int64_t)b * row;
    const float* rb = r + (int64_t)b * row;
    const float* ob = cand + (int64_t)e * row;
    float part = 0.f;
    for (int k = threadIdx.x; k < dc; k += KT) {
      float ure = sb[k] * rb[k] - sb[dc + k] * rb[dc + k];
      float uim = sb[dc + k] * rb[k] + sb[k] * rb[dc + k];
      part += ure * ob[k] + uim * ob[dc + k];
    }
    float psi = block_reduce_sum(part, lds);
    if (threadIdx.x == 0) scores[be] = psi;
  }
}

// --------------------------------------------------------------- SGNS

template <int KPT>
__global__ void k_w2v_step(const float* __restrict__ ctr, const float* __restrict__ ctx,
                           const float* __restrict__ neg, float* __restrict__ dctr,
                           float* __restrict__ dctx, float* __restrict__ dneg,
                           float* __restrict__ loss, int B, int N, int D, float lr, float eps) {
  __shared__ float lds[KT / 64];
  const int row = D << 1;
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    const float* cb = ctr + (int64_t)b * row;
    float c_emb[KPT], a_c[KPT];
#pragma unroll
    for (int i = 0; i < KPT; ++i) {
      int k = threadIdx.x + i * KT;
      c_emb[i] = k < D ? cb[k] : 0.f;
      a_c[i] = 0.f;
    }

    float lsum = 0.f;
    for (int j = 0; j <= N; ++j) {
      const float* xb = (j == 0) ? ctx + (int64_t)b * row
                                 : neg + ((int64_t)b * N + (j - 1)) * row;
      float y = (j == 0) ? 1.f : -1.f;
      float part = 0.f;
      float x_emb[KPT];
#pragma unroll
      for (int i = 0; i < KPT; ++i) {
        int k = threadIdx.x + i * KT;
        x_emb[i] = k < D ? xb[k] : 0.f;
        part += c_emb[i] * x_emb[i];
      }
      float dot = block_reduce_sum(part, lds);
      float g = -y * sigmoidf(-y * dot);
      if (threadIdx.x == 0) lsum += softplusf(-y * dot);
      float* dxb = (j == 0) ? dctx + (int64_t)b * row
                            : dneg + ((int64_t)b * N + (j - 1)) * row;
#pragma unroll
      for (int i = 0; i < KPT; ++i) {
        int k = threadIdx.x + i * KT;
        if (k >= D) continue;
        a_c[i] += g * x_emb[i];
        float gx = g * c_emb[i];
        float G = xb[D + k] + gx * gx;
        dxb[k] = -lr * gx * __frsqrt_rn(G + eps);
        dxb[D + k] = gx * gx;
      }
    }
    float* dcb = dctr + (int64_t)b * row;
#pragma unroll
    for (int i = 0; i < KPT; ++i) {
      int k = threadIdx.x + i * KT;
      if (k >= D) continue;
      float G = cb[D + k] + a_c[i] * a_c[i];
      dcb[k] = -lr * a_c[i] * __frsqrt_rn(G + eps);
      dcb[D + k] = a_c[i] * a_c[i];
    }
    if (threadIdx.x == 0) loss[b] = lsum;
  }
}

// --------------------------------------------------------------- MF

__global__ void k_mf_step(const float* __restrict__ w, const float* __restrict__ h,
                          const float* __restrict__ x, float* __restrict__ dw,
                          float* __restrict__ dh, float* __restrict__ loss, int B, int R,
                          float lr, float lambda, float eps) {
  __shared__ float lds[KT / 64];
  const int row = R << 1;
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    const float* wb = w + (int64_t)b * row;
    const float* hb = h + (int64_t)b * row;
    float part = 0.f;
    for (int k = threadIdx.x; k < R; k += KT) part += wb[k] * hb[k];
    float pred = block_reduce_sum(part, lds);
    float e = x[b] - pred;
    if (threadIdx.x == 0) loss[b] = e * e;
    float* dwb = dw + (int64_t)b * row;
    float* dhb = dh + (int64_t)b * row;
    for (int k = threadIdx.x; k < R; k += KT) {
      float gw = -2.f * e * hb[k] + 2.f * lambda * wb[k];
      float gh = -2.f * e * wb[k] + 2.f * lambda * hb[k];
      float Gw = wb[R + k] + gw * gw;
      float Gh = hb[R + k] + gh * gh;
      dwb[k] = -lr * gw * __frsqrt_rn(Gw + eps);
      dwb[R + k] = gw * gw;
      dhb[k] = -lr * gh * __frsqrt_rn(Gh + eps);
      dhb[R + k] = gh * gh;
    }
  }
}

// --------------------------------------------------------------- MF loss

__global__ void k_mf_loss(const float* __restrict__ w, const float* __restrict__ h,
                          const float* __restrict__ x, float* __restrict__ out2, int B, int R,
                          float lambda) {
  __shared__ float lds[2][KT / 64];
  const int row = R << 1;
  float se = 0.f, reg = 0.f;
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    const float* wb = w + (int64_t)b * row;
    const float* hb = h + (int64_t)b * row;
    float part = 0.f, nrm = 0.f;
    for (int k = threadIdx.x; k < R; k += KT) {
      float wv = wb[k], hv = hb[k];
      part += wv * hv;
      nrm += wv * wv + hv * hv;
    }
    float pred = block_reduce_sum(part, lds[0]);
    float n2 = block_reduce_sum(nrm, lds[1]);
    if (threadIdx.x == 0) {
      float e = x[b] - pred;
      se += e * e;
      reg += lambda * n2;
    }
  }
  if (threadIdx.x == 0) {
    atomicAdd(&out2[0], se);
    atomicAdd(&out2[1], reg);
  }
}

void mf_loss_gpu(const float* w, const float* h, const float* x, float* out2, int B, int R,
                 float lambda, void* stream) {
  if (B < 1) return;
  int g = B > 16384 ? 16384 : B;
  hipLaunchKernelGGL(k_mf_loss, dim3(g), dim3(KT), 0, (hipStream_t)stream, w, h, x,
                     out2, B, R, lambda);
}

// --------------------------------------------------------------- alias draw

__device__ __host__ inline uint64_t pcg_hash64(uint64_t x) {
  x ^= x >> 33; x *= 0xff51afd7ed558ccdULL;
  x ^= x >> 33; x *= 0xc4ceb9fe1a85ec53ULL;
  x ^= x >> 33;
  return x;
}

__global__ void k_alias_draw(const float* __restrict__ prob, const int32_t* __restrict__ alias,
                             int64_t n, uint64_t seed, int64_t N, int64_t* __restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < N; i += stride) {
    uint64_t h = pcg_hash64(seed ^ (uint64_t)i * 0x9e3779b97f4a7c15ULL);
    int64_t slot = (int64_t)(h % (uint64_t)n);
    float u = (float)((h >> 40) & 0xffffff) * (1.0f / 16777216.0f);
    out[i] = (u < prob[slot]) ? slot : (int64_t)alias[slot];
  }
}

// --------------------------------------------------------------- launchers

static inline int grid_for(int64_t n) {
  int64_t g = n < 1 ? 1 : n;
  return (int)(g > 16384 ? 16384 : g);
}

void kge_complex_step_gpu(const float* s, const float* r, const float* o, const float* neg,
                          float* ds, float* dr, float* do_, float* dneg, float* loss, int B,
                          int N, int D, float lr, float eps, void* stream) {
  int dc = D >> 1;
  dim3 g(grid_for(B)), t(KT);
  auto st = (hipStream_t)stream;
#define LAUNCH(KPT) \
  hipLaunchKernelGGL(k_kge_step<KPT>, g, t, 0, st, s, r, o, neg, ds, dr, do_, dneg, loss, B, N, \
                     D, lr, eps)
  if (dc <= KT) LAUNCH(1);
  else if (dc <= 2 * KT) LAUNCH(2);
  else if (dc <= 4 * KT) LAUNCH(4);
  else LAUNCH(8);
#undef LAUNCH
}
void kge_complex_score_gpu(const float* s, const float* r, const float* cand, float* scores,
                           int B, int E, int D, void* stream) {
  hipLaunchKernelGGL(k_kge_score, dim3(grid_for((int64_t)B * E)), dim3(KT), 0,
                     (hipStream_t)stream, s, r, cand, scores, B, E, D);
}
void w2v_sgns_step_gpu(const float* ctr, const float* ctx, const float* neg, float* dctr,
                       float* dctx, float* dneg, float* loss, int B, int N, int D, float lr,
                       float eps, void* stream) {
  dim3 g(grid_for(B)), t(KT);
  auto st = (hipStream_t)stream;
#define LAUNCH(KPT) \
  hipLaunchKernelGGL(k_w2v_step<KPT>, g, t, 0, st, ctr, ctx, neg, dctr, dctx, dneg, loss, B, N, \
                     D, lr, eps)
  if (D <= KT) LAUNCH(1);
  else if (D <= 2 * KT) LAUNCH(2);
  else if (D <= 4 * KT) LAUNCH(4);
  else LAUNCH(8);
#undef LAUNCH
}
void mf_update_step_gpu(const float* w, const float* h, const float* x, float* dw, float* dh,
                        float* loss, int B, int R, float lr, float lambda, float eps,
                        void* stream) {
  hipLaunchKernelGGL(k_mf_step, dim3(grid_for(B)), dim3(KT), 0, (hipStream_t)stream, w, h, x, dw,
                     dh, loss, B, R, lr, lambda, eps);
}

}  // namespace adapm

namespace adapm {
void alias_draw_gpu(const float* prob, const int32_t* alias, int64_t n, uint64_t seed,
                    int64_t N, int64_t* out, void* stream) {
  if (N == 0) return;
  int blocks = (int)std::min<int64_t>((N + 255) / 256, 4096);
  hipLaunchKernelGGL(k_alias_draw, dim3(blocks), dim3(256), 0, (hipStream_t)stream, prob, alias,
                     n, seed, N, out);
}
}  // namespace adapm

namespace adapm {

// RESCAL step: one workgroup per positive triple; R staged in LDS
// (dim <= 128 -> 64 KiB). See kernels.h for the math factorization.
__global__ void k_rescal_step(const float* __restrict__ s, const float* __restrict__ r,
                              const float* __restrict__ o, const float* __restrict__ neg,
                              float* __restrict__ ds, float* __restrict__ drl,
                              float* __restrict__ do_, float* __restrict__ dneg,
                              float* __restrict__ loss, int B, int N, int D, float lr,
                              float eps) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  float* Rm = smem;            // D*D
  float* u = Rm + D * D;       // D
  float* w = u + D;            // D
  float* es = w + D;           // D
  float* red = es + D;         // KT/64 reduction scratch
  const int erow = 2 * D;
  const int64_t rrow = 2LL * D * D;

  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    const float* sb = s + (int64_t)b * erow;
    const float* rb = r + (int64_t)b * rrow;
    for (int i = threadIdx.x; i < D * D; i += KT) Rm[i] = rb[i];
    for (int i = threadIdx.x; i < D; i += KT) {
      es[i] = sb[i];
      w[i] = 0.f;
    }
    __syncthreads();
    // u = R^T e_s: thread j computes column j
    for (int j = threadIdx.x; j < D; j += KT) {
      float a = 0.f;
      for (int i = 0; i < D; ++i) a += Rm[i * D + j] * es[i];
      u[j] = a;
    }
    __syncthreads();

    float lsum = 0.f;
    for (int j = 0; j <= N; ++j) {
      const float* ob = (j == 0) ? o + (int64_t)b * erow
                                 : neg + ((int64_t)b * N + j - 1) * erow;
      float* dob = (j == 0) ? do_ + (int64_t)b * erow
                            : dneg + ((int64_t)b * N + j - 1) * erow;
      float y = (j == 0) ? 1.f : -1.f;
      float part = 0.f;
      for (int k = threadIdx.x; k < D; k += KT) part += u[k] * ob[k];
      float dot = block_reduce_sum(part, red);
      float c = -y * sigmoidf(-y * dot);
      if (threadIdx.x == 0) lsum += softplusf(-y * dot);
      for (int k = threadIdx.x; k < D; k += KT) {
        w[k] += c * ob[k];  // single writer per k: thread-private index
        float g = c * u[k];
        dob[k] = -lr * g * __frsqrt_rn(ob[D + k] + g * g + eps);
        dob[D + k] = g * g;
      }
      __syncthreads();
    }
    if (threadIdx.x == 0) loss[b] = lsum;

    // de_s = R w
    float* dsb = ds + (int64_t)b * erow;
    for (int i = threadIdx.x; i < D; i += KT) {
      float a = 0.f;
      for (int k = 0; k < D; ++k) a += Rm[i * D + k] * w[k];
      float g = a;
      dsb[i] = -lr * g * __frsqrt_rn(sb[D + i] + g * g + eps);
      dsb[D + i] = g * g;
    }
    // dR = e_s w^T
    float* drb = drl + (int64_t)b * rrow;
    for (int idx = threadIdx.x; idx < D * D; idx += KT) {
      int i = idx / D, k = idx % D;
      float g = es[i] * w[k];
      drb[idx] = -lr * g * __frsqrt_rn(rb[(int64_t)D * D + idx] + g * g + eps);
      drb[(int64_t)D * D + idx] = g * g;
    }
    __syncthreads();
  }
}

void rescal_step_gpu(const float* s, const float* r, const float* o, const float* neg,
                     float* ds, float* drl, float* do_, float* dneg, float* loss, int B, int N,
                     int D, float lr, float eps, void* stream) {
  size_t smem = (size_t)(D * D + 3 * D + KT / 64) * sizeof(float);
  int blocks = (int)std::min<int64_t>(B, 8192);
  hipLaunchKernelGGL(k_rescal_step, dim3(blocks), dim3(KT), smem, (hipStream_t)stream, s, r, o,
                     neg, ds, drl, do_, dneg, loss, B, N, D, lr, eps);
}

}  // namespace adapm

namespace adapm {

// Row resolution for the fused slab-direct kernels. world >= 1: identity
// layout, key k lives at (k/world)*plen (the world==1 single-rank fast
// path with zero host work). world == 0: the "keys" arrays carry
// precomputed float OFFSETS into the slab (the world>1 / relocated-layout
// path: the host metadata pass resolves each key's slot and compacts
// all-local samples; remote ones go through the classic pull/push path).
__device__ inline float* row_at(float* slab, int64_t v, int32_t plen, int world) {
  return world ? slab + (v / world) * (int64_t)plen : slab + v;
}

// Fused ComplEx step — see kernels.h. Structure mirrors k_kge_step but
// rows come from / return to the slab itself.
template <int KPT>
__global__ void k_kge_step_fused(float* __restrict__ slab, const int64_t* __restrict__ keys_s,
                                 const int64_t* __restrict__ keys_r,
                                 const int64_t* __restrict__ keys_o,
                                 const int64_t* __restrict__ keys_neg, float* __restrict__ loss,
                                 int B, int N, int D, int32_t plen, int world, int rank,
                                 float lr, float eps) {
  __shared__ float lds[KT / 64];
  const int dc = D >> 1;
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    const float* sb = row_at(slab, keys_s[b], plen, world);
    const float* rb = row_at(slab, keys_r[b], plen, world);

    float s_re[KPT], s_im[KPT], r_re[KPT], r_im[KPT];
    float a_sre[KPT], a_sim[KPT], a_rre[KPT], a_rim[KPT];
    float u_re[KPT], u_im[KPT];
#pragma unroll
    for (int i = 0; i < KPT; ++i) {
      int k = threadIdx.x + i * KT;
      a_sre[i] = a_sim[i] = a_rre[i] = a_rim[i] = 0.f;
      if (k < dc) {
        s_re[i] = sb[k];
        s_im[i] = sb[dc + k];
        r_re[i] = rb[k];
        r_im[i] = rb[dc + k];
        u_re[i] = s_re[i] * r_re[i] - s_im[i] * r_im[i];
        u_im[i] = s_im[i] * r_re[i] + s_re[i] * r_im[i];
      } else {
        s_re[i] = s_im[i] = r_re[i] = r_im[i] = u_re[i] = u_im[i] = 0.f;
      }
    }

    float lsum = 0.f;
    for (int j = 0; j <= N; ++j) {
      int64_t ok = (j == 0) ? keys_o[b] : keys_neg[(int64_t)b * N + (j - 1)];
      float* ob = row_at(slab, ok, plen, world);
      float y = (j == 0) ? 1.f : -1.f;
      float part = 0.f;
      float o_re[KPT], o_im[KPT];
#pragma unroll
      for (int i = 0; i < KPT; ++i) {
        int k = threadIdx.x + i * KT;
        o_re[i] = k < dc ? ob[k] : 0.f;
        o_im[i] = k < dc ? ob[dc + k] : 0.f;
        part += u_re[i] * o_re[i] + u_im[i] * o_im[i];
      }
      float psi = block_reduce_sum(part, lds);
      float c = -y * sigmoidf(-y * psi);
      if (threadIdx.x == 0) lsum += softplusf(-y * psi);

#pragma unroll
      for (int i = 0; i < KPT; ++i) {
        int k = threadIdx.x + i * KT;
        if (k >= dc) continue;
        a_sre[i] += c * (r_re[i] * o_re[i] + r_im[i] * o_im[i]);
        a_sim[i] += c * (r_re[i] * o_im[i] - r_im[i] * o_re[i]);
        a_rre[i] += c * (s_re[i] * o_re[i] + s_im[i] * o_im[i]);
        a_rim[i] += c * (s_re[i] * o_im[i] - s_im[i] * o_re[i]);
        float g_re = c * u_re[i];
        float g_im = c * u_im[i];
        float G_re = ob[D + k] + g_re * g_re;
        float G_im = ob[D + dc + k] + g_im * g_im;
        atomicAdd(&ob[k], -lr * g_re * __frsqrt_rn(G_re + eps));
        atomicAdd(&ob[dc + k], -lr * g_im * __frsqrt_rn(G_im + eps));
        atomicAdd(&ob[D + k], g_re * g_re);
        atomicAdd(&ob[D + dc + k], g_im * g_im);
      }
    }
    float* dsb = const_cast<float*>(sb);
    float* drb = const_cast<float*>(rb);
#pragma unroll
    for (int i = 0; i < KPT; ++i) {
      int k = threadIdx.x + i * KT;
      if (k >= dc) continue;
      float Gsr = sb[D + k] + a_sre[i] * a_sre[i];
      float Gsi = sb[D + dc + k] + a_sim[i] * a_sim[i];
      atomicAdd(&dsb[k], -lr * a_sre[i] * __frsqrt_rn(Gsr + eps));
      atomicAdd(&dsb[dc + k], -lr * a_sim[i] * __frsqrt_rn(Gsi + eps));
      atomicAdd(&dsb[D + k], a_sre[i] * a_sre[i]);
      atomicAdd(&dsb[D + dc + k], a_sim[i] * a_sim[i]);
      float Grr = rb[D + k] + a_rre[i] * a_rre[i];
      float Gri = rb[D + dc + k] + a_rim[i] * a_rim[i];
      atomicAdd(&drb[k], -lr * a_rre[i] * __frsqrt_rn(Grr + eps));
      atomicAdd(&drb[dc + k], -lr * a_rim[i] * __frsqrt_rn(Gri + eps));
      atomicAdd(&drb[D + k], a_rre[i] * a_rre[i]);
      atomicAdd(&drb[D + dc + k], a_rim[i] * a_rim[i]);
    }
    if (threadIdx.x == 0) loss[b] = lsum;
  }
}

void kge_complex_step_fused_gpu(float* slab, const int64_t* keys_s, const int64_t* keys_r,
                                const int64_t* keys_o, const int64_t* keys_neg, float* loss,
                                int B, int N, int D, int32_t plen, int world, int rank,
                                float lr, float eps, void* stream) {
  int dc = D >> 1;
  dim3 g((unsigned)std::min<int64_t>(B, 16384)), t(KT);
  auto st = (hipStream_t)stream;
#define LAUNCHF(KPT) \
  hipLaunchKernelGGL(k_kge_step_fused<KPT>, g, t, 0, st, slab, keys_s, keys_r, keys_o, \
                     keys_neg, loss, B, N, D, plen, world, rank, lr, eps)
  if (dc <= KT) LAUNCHF(1);
  else if (dc <= 2 * KT) LAUNCHF(2);
  else if (dc <= 4 * KT) LAUNCHF(4);
  else LAUNCHF(8);
#undef LAUNCHF
}

}  // namespace adapm

namespace adapm {

// Fused SGNS: mirror of k_w2v_step reading rows straight from the slab
// at identity offsets and atomically applying the AdaGrad updates.
template <int KPT>
__global__ void k_w2v_step_fused(float* __restrict__ slab, const int64_t* __restrict__ keys_ctr,
                                 const int64_t* __restrict__ keys_ctx,
                                 const int64_t* __restrict__ keys_neg, float* __restrict__ loss,
                                 int B, int N, int D, int32_t plen, int world, float lr,
                                 float eps) {
  __shared__ float lds[KT / 64];
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    float* cb = row_at(slab, keys_ctr[b], plen, world);
    float c_emb[KPT], a_c[KPT];
#pragma unroll
    for (int i = 0; i < KPT; ++i) {
      int k = threadIdx.x + i * KT;
      c_emb[i] = k < D ? cb[k] : 0.f;
      a_c[i] = 0.f;
    }
    float lsum = 0.f;
    for (int j = 0; j <= N; ++j) {
      int64_t xk = (j == 0) ? keys_ctx[b] : keys_neg[(int64_t)b * N + (j - 1)];
      float* xb = row_at(slab, xk, plen, world);
      float y = (j == 0) ? 1.f : -1.f;
      float part = 0.f;
      float x_emb[KPT];
#pragma unroll
      for (int i = 0; i < KPT; ++i) {
        int k = threadIdx.x + i * KT;
        x_emb[i] = k < D ? xb[k] : 0.f;
        part += c_emb[i] * x_emb[i];
      }
      float dot = block_reduce_sum(part, lds);
      float g = -y * sigmoidf(-y * dot);
      if (threadIdx.x == 0) lsum += softplusf(-y * dot);
#pragma unroll
      for (int i = 0; i < KPT; ++i) {
        int k = threadIdx.x + i * KT;
        if (k >= D) continue;
        a_c[i] += g * x_emb[i];
        float gx = g * c_emb[i];
        float G = xb[D + k] + gx * gx;
        atomicAdd(&xb[k], -lr * gx * __frsqrt_rn(G + eps));
        atomicAdd(&xb[D + k], gx * gx);
      }
    }
#pragma unroll
    for (int i = 0; i < KPT; ++i) {
      int k = threadIdx.x + i * KT;
      if (k >= D) continue;
      float G = cb[D + k] + a_c[i] * a_c[i];
      atomicAdd(&cb[k], -lr * a_c[i] * __frsqrt_rn(G + eps));
      atomicAdd(&cb[D + k], a_c[i] * a_c[i]);
    }
    if (threadIdx.x == 0) loss[b] = lsum;
  }
}

// Fused MF: w/h row values are read into registers before any atomic
// write in the same lane, so the NZSL+L2 gradients use a consistent
// pre-update snapshot per nonzero (duplicates across the batch hogwild).
__global__ void k_mf_step_fused(float* __restrict__ slab, const int64_t* __restrict__ keys_w,
                                const int64_t* __restrict__ keys_h, const float* __restrict__ x,
                                float* __restrict__ loss, int B, int R, int32_t plen, int world,
                                float lr, float lambda, float eps) {
  __shared__ float lds[KT / 64];
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    float* wb = row_at(slab, keys_w[b], plen, world);
    float* hb = row_at(slab, keys_h[b], plen, world);
    float part = 0.f;
    for (int k = threadIdx.x; k < R; k += KT) part += wb[k] * hb[k];
    float pred = block_reduce_sum(part, lds);
    float e = x[b] - pred;
    if (threadIdx.x == 0) loss[b] = e * e;
    for (int k = threadIdx.x; k < R; k += KT) {
      float wv = wb[k], hv = hb[k];
      float gw = -2.f * e * hv + 2.f * lambda * wv;
      float gh = -2.f * e * wv + 2.f * lambda * hv;
      float Gw = wb[R + k] + gw * gw;
      float Gh = hb[R + k] + gh * gh;
      atomicAdd(&wb[k], -lr * gw * __frsqrt_rn(Gw + eps));
      atomicAdd(&wb[R + k], gw * gw);
      atomicAdd(&hb[k], -lr * gh * __frsqrt_rn(Gh + eps));
      atomicAdd(&hb[R + k], gh * gh);
    }
  }
}

void w2v_sgns_step_fused_gpu(float* slab, const int64_t* keys_ctr, const int64_t* keys_ctx,
                             const int64_t* keys_neg, float* loss, int B, int N, int D,
                             int32_t plen, int world, float lr, float eps, void* stream) {
  dim3 g((unsigned)std::min<int64_t>(B < 1 ? 1 : B, 16384)), t(KT);
  auto st = (hipStream_t)stream;
#define LAUNCHW(KPT) \
  hipLaunchKernelGGL(k_w2v_step_fused<KPT>, g, t, 0, st, slab, keys_ctr, keys_ctx, keys_neg, \
                     loss, B, N, D, plen, world, lr, eps)
  if (D <= KT) LAUNCHW(1);
  else if (D <= 2 * KT) LAUNCHW(2);
  else if (D <= 4 * KT) LAUNCHW(4);
  else LAUNCHW(8);
#undef LAUNCHW
}

void mf_update_step_fused_gpu(float* slab, const int64_t* keys_w, const int64_t* keys_h,
                              const float* x, float* loss, int B, int R, int32_t plen,
                              int world, float lr, float lambda, float eps, void* stream) {
  hipLaunchKernelGGL(k_mf_step_fused, dim3((unsigned)std::min<int64_t>(B < 1 ? 1 : B, 16384)),
                     dim3(KT), 0, (hipStream_t)stream, slab, keys_w, keys_h, x, loss, B, R,
                     plen, world, lr, lambda, eps);
}

}  // namespace adapm
