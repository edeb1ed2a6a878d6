// MFMA (matrix-core) kernels for the GEMM-shaped compute paths:
// KGE full-entity eval scoring and grouped RESCAL products.
//
// gfx950 keeps fp32-input MFMA (v_mfma_f32_16x16x4_f32): exact f32 at
// the f32 vector rate, but ~2.4-2.8x a VALU f32 GEMM in practice (one
// VGPR per operand, accumulators in AGPRs, VALU free for the epilogue).
// Shapes here are fp32 training state, so this is the right instrument
// (no xf32 on gfx950; bf16 would change numerics).
//
// Tiling: one workgroup = 4 waves; each wave owns one 16x16 output tile
// -> workgroup tile 16(M) x 64(N). A-tile (16 x KC) is staged in LDS
// once per workgroup; B-tiles (64 x KC) staged per wave. KC = 32 with
// +1 padding against bank conflicts. Lane mapping for
// v_mfma_f32_16x16x4_f32: A[l&15][l>>4], B[l>>4][l&15], C/D col=l&15,
// row=(l>>4)*4+reg.
#include <hip/hip_runtime.h>
#include <algorithm>

#include "kernels.h"

namespace adapm {

#define MT 256  // threads per workgroup (4 waves)
using f32x4 = __attribute__((ext_vector_type(4))) float;

// scores[b][e] = sum_k Q[b][k] * C[e][k]
// Q: [B][qstride] dense query rows (first K used)
// C: [E][cstride] candidate rows (first K used)
// Workgroup tile: 16 queries x 256 candidates — each of the 4 waves
// holds FOUR 16x16 accumulators over its 64-candidate strip, so every
// staged A element feeds 4 MFMAs (the single-accumulator version
// measured 29 TF/s; the issue-rate cap for f32-in MFMA is 155).
__global__ void k_mfma_scores(const float* __restrict__ Q, const float* __restrict__ C,
                              float* __restrict__ out, int B, int E, int K, int qstride,
                              int cstride) {
  constexpr int KC = 16;  // keeps LDS ~17 KB/WG so ~8 workgroups co-reside per CU
  __shared__ float lq[16][KC + 1];
  __shared__ float lc[256][KC + 1];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  int mtiles = (B + 15) / 16;
  int ntiles = (E + 255) / 256;
  for (int tile = blockIdx.x; tile < mtiles * ntiles; tile += gridDim.x) {
    int m0 = (tile % mtiles) * 16;
    int n0 = (tile / mtiles) * 256;
    f32x4 acc[4] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f},
                    {0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
    for (int k0 = 0; k0 < K; k0 += KC) {
      int kc = min(KC, K - k0);
      // stage A (16 x kc): 256 threads cover 512 floats -> 2 each
      for (int idx = threadIdx.x; idx < 16 * KC; idx += MT) {
        int r = idx / KC, c = idx % KC;
        lq[r][c] = (m0 + r < B && c < kc) ? Q[(int64_t)(m0 + r) * qstride + k0 + c] : 0.f;
      }
      // stage B (256 x kc): 8192 floats -> 32 each
      for (int idx = threadIdx.x; idx < 256 * KC; idx += MT) {
        int r = idx / KC, c = idx % KC;
        lc[r][c] = (n0 + r < E && c < kc) ? C[(int64_t)(n0 + r) * cstride + k0 + c] : 0.f;
      }
      __syncthreads();
      const int arow = lane & 15;          // A row (query within tile)
      const int kk = lane >> 4;            // k within the 4-slice
#pragma unroll
      for (int ks = 0; ks < KC; ks += 4) {
        float a = lq[arow][ks + kk];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          int brow = (wave << 6) | (j << 4) | (lane & 15);
          acc[j] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, lc[brow][ks + kk], acc[j], 0, 0, 0);
        }
      }
      __syncthreads();
    }
    // write: col = lane&15 within sub-tile j, row = (lane>>4)*4+reg
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + (wave << 6) + (j << 4) + (lane & 15);
      if (col >= E) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + ((lane >> 4) << 2) + r;
        if (row < B) out[(int64_t)row * E + col] = acc[j][r];
      }
    }
  }
}

// Q[b][k<dc] = s_re*r_re - s_im*r_im ; Q[b][dc+k] = s_im*r_re + s_re*r_im
__global__ void k_build_query(const float* __restrict__ s, const float* __restrict__ r,
                              float* __restrict__ q, int B, int D) {
  const int dc = D >> 1;
  const int row = D << 1;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t total = (int64_t)B * dc;
  for (; i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int b = (int)(i / dc);
    int k = (int)(i % dc);
    const float* sb = s + (int64_t)b * row;
    const float* rb = r + (int64_t)b * row;
    float sre = sb[k], sim = sb[dc + k], rre = rb[k], rim = rb[dc + k];
    q[(int64_t)b * D + k] = sre * rre - sim * rim;
    q[(int64_t)b * D + dc + k] = sim * rre + sre * rim;
  }
}

void kge_complex_score_mfma_gpu(const float* s, const float* r, const float* cand,
                                float* scores, float* qbuf, int B, int E, int D,
                                void* stream) {
  auto st = (hipStream_t)stream;
  int dc = D >> 1;
  {
    int64_t total = (int64_t)B * dc;
    int blocks = (int)std::min<int64_t>((total + 255) / 256, 4096);
    hipLaunchKernelGGL(k_build_query, dim3(blocks), dim3(256), 0, st, s, r, qbuf, B, D);
  }
  int mtiles = (B + 15) / 16, ntiles = (E + 255) / 256;
  int blocks = (int)std::min<int64_t>((int64_t)mtiles * ntiles, 8192);
  // candidate rows are [emb(D) | accum(D)]: stride 2D, first D used; the
  // query buffer is dense D
  hipLaunchKernelGGL(k_mfma_scores, dim3(blocks), dim3(MT), 0, st, qbuf, cand, scores, B, E, D,
                     D, D << 1);
}

// ---------------------------------------------------------------- grouped GEMMs
//
// Batched varying-M GEMMs for grouped RESCAL (triples sorted by
// relation; group g spans rows [starts[g], starts[g+1]) and uses
// relation matrix g):
//   mode 0:  U  = S  @ R      (M=Bg, K=D, N=D)   u_b = R^T e_s  per row
//   mode 1:  dS = W  @ R^T    (M=Bg, K=D, N=D)
//   mode 2:  dR = S^T @ W     (M=D,  K=Bg, N=D)
// R matrices: Rm + (int64)g * rstride (row-major D x D).
// S/W/U/dS: [B][stride] rows (first D cols used).
__global__ void k_mfma_grouped(const float* __restrict__ Sm, const float* __restrict__ Wm,
                               const float* __restrict__ Rm, float* __restrict__ Out,
                               const int* __restrict__ starts, int G, int D, int sstride,
                               int wstride, int64_t rstride, int out_rstride,
                               int64_t ostride, int mode) {
  constexpr int KC = 32;
  __shared__ float la[16][KC + 1];
  __shared__ float lb[64][KC + 1];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int ntiles = (D + 63) / 64;

  // flatten (group, mtile, ntile) over a grid-stride loop. M depends on
  // the group in modes 0/1 (Bg) and is D in mode 2.
  // Precompute per-group tile counts on the fly (G is small).
  int total = 0;
  for (int g = 0; g < G; ++g) {
    int Bg = starts[g + 1] - starts[g];
    int M = (mode == 2) ? D : Bg;
    total += ((M + 15) / 16) * ntiles;
  }
  for (int t = blockIdx.x; t < total; t += gridDim.x) {
    // locate the group
    int g = 0, base = 0;
    for (;; ++g) {
      int Bg = starts[g + 1] - starts[g];
      int M = (mode == 2) ? D : Bg;
      int nt = ((M + 15) / 16) * ntiles;
      if (t < base + nt) break;
      base += nt;
    }
    int Bg = starts[g + 1] - starts[g];
    int M = (mode == 2) ? D : Bg;
    int K = (mode == 2) ? Bg : D;
    int lt = t - base;
    int m0 = (lt % ((M + 15) / 16)) * 16;
    int n0 = (lt / ((M + 15) / 16)) * 64;
    const float* R = Rm + (int64_t)g * rstride;
    const int s0 = starts[g];

    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    for (int k0 = 0; k0 < K; k0 += KC) {
      int kc = min(KC, K - k0);
      // stage A tile (16 x kc)
      for (int idx = threadIdx.x; idx < 16 * KC; idx += MT) {
        int rr = idx / KC, cc = idx % KC;
        float v = 0.f;
        if (cc < kc && m0 + rr < M) {
          if (mode == 0)       v = Sm[(int64_t)(s0 + m0 + rr) * sstride + k0 + cc];
          else if (mode == 1)  v = Wm[(int64_t)(s0 + m0 + rr) * wstride + k0 + cc];
          else                 v = Sm[(int64_t)(s0 + k0 + cc) * sstride + m0 + rr];  // S^T
        }
        la[rr][cc] = v;
      }
      // stage B tile (64 x kc): B[k][n] consumed as lb[n][k]
      for (int idx = threadIdx.x; idx < 64 * KC; idx += MT) {
        int rr = idx / KC, cc = idx % KC;  // rr = n offset, cc = k offset
        float v = 0.f;
        if (cc < kc && n0 + rr < D) {
          if (mode == 0)       v = R[(int64_t)(k0 + cc) * D + n0 + rr];      // R[k][n]
          else if (mode == 1)  v = R[(int64_t)(n0 + rr) * D + k0 + cc];      // R^T[k][n]
          else                 v = Wm[(int64_t)(s0 + k0 + cc) * wstride + n0 + rr];  // W[k][n]
        }
        lb[rr][cc] = v;
      }
      __syncthreads();
      const int arow = lane & 15;
      const int brow = (wave << 4) | (lane & 15);
      const int kk = lane >> 4;
#pragma unroll
      for (int ks = 0; ks < KC; ks += 4) {
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(la[arow][ks + kk], lb[brow][ks + kk], acc,
                                                   0, 0, 0);
      }
      __syncthreads();
    }
    int col = n0 + (wave << 4) + (lane & 15);
    if (col < D) {
#pragma unroll
      for (int rg = 0; rg < 4; ++rg) {
        int row = m0 + ((lane >> 4) << 2) + rg;
        if (row < M) {
          if (mode == 2)
            Out[(int64_t)g * ostride + (int64_t)row * D + col] = acc[rg];
          else
            Out[(int64_t)(s0 + row) * out_rstride + col] = acc[rg];
        }
      }
    }
  }
}

void mfma_grouped_gemm_gpu(const float* S, const float* W, const float* R, float* out,
                           const int* starts, int G, int D, int sstride, int wstride,
                           int64_t rstride, int out_rstride, int64_t ostride, int mode,
                           int total_tiles, void* stream) {
  int blocks = std::min(total_tiles, 8192);
  if (blocks < 1) return;
  hipLaunchKernelGGL(k_mfma_grouped, dim3(blocks), dim3(MT), 0, (hipStream_t)stream, S, W, R,
                     out, starts, G, D, sstride, wstride, rstride, out_rstride, ostride, mode);
}

// per-triple middle phase of the grouped RESCAL step: scores, object
// grads (fused AdaGrad), W accumulation. U was produced by the mode-0
// grouped GEMM (u_b = R_g^T e_s).
__global__ void k_rescal_mid(const float* __restrict__ U, const float* __restrict__ o,
                             const float* __restrict__ neg, float* __restrict__ do_,
                             float* __restrict__ dneg, float* __restrict__ Wm,
                             float* __restrict__ loss, int B, int N, int D, float lr,
                             float eps) {
  __shared__ float red[MT / 64];
  const int erow = 2 * D;
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    const float* ub = U + (int64_t)b * D;
    float* wb = Wm + (int64_t)b * D;
    for (int k = threadIdx.x; k < D; k += MT) wb[k] = 0.f;
    __syncthreads();
    float lsum = 0.f;
    for (int j = 0; j <= N; ++j) {
      const float* ob = (j == 0) ? o + (int64_t)b * erow
                                 : neg + ((int64_t)b * N + j - 1) * erow;
      float* dob = (j == 0) ? do_ + (int64_t)b * erow
                            : dneg + ((int64_t)b * N + j - 1) * erow;
      float y = (j == 0) ? 1.f : -1.f;
      float part = 0.f;
      for (int k = threadIdx.x; k < D; k += MT) part += ub[k] * ob[k];
      // block reduce (wave shfl + LDS hop)
      for (int off = 32; off > 0; off >>= 1) part += __shfl_xor(part, off, 64);
      int wv = threadIdx.x >> 6, ln = threadIdx.x & 63;
      if (ln == 0) red[wv] = part;
      __syncthreads();
      float dot = (threadIdx.x < MT / 64) ? red[threadIdx.x] : 0.f;
      if (threadIdx.x < 64)
        for (int off = 2; off > 0; off >>= 1) dot += __shfl_xor(dot, off, 64);
      if (threadIdx.x == 0) red[0] = dot;
      __syncthreads();
      dot = red[0];
      __syncthreads();
      float c = -y / (1.f + __expf(y * dot));
      if (threadIdx.x == 0)
        lsum += (-y * dot > 20.f) ? -y * dot : __logf(1.f + __expf(-y * dot));
      for (int k = threadIdx.x; k < D; k += MT) {
        wb[k] += c * ob[k];
        float g = c * ub[k];
        dob[k] = -lr * g * __frsqrt_rn(ob[D + k] + g * g + eps);
        dob[D + k] = g * g;
      }
      __syncthreads();
    }
    if (threadIdx.x == 0) loss[b] = lsum;
  }
}

void rescal_mid_gpu(const float* U, const float* o, const float* neg, float* do_, float* dneg,
                    float* Wm, float* loss, int B, int N, int D, float lr, float eps,
                    void* stream) {
  if (B < 1) return;
  hipLaunchKernelGGL(k_rescal_mid, dim3(std::min(B, 8192)), dim3(MT), 0, (hipStream_t)stream,
                     U, o, neg, do_, dneg, Wm, loss, B, N, D, lr, eps);
}

// AdaGrad epilogues: transform a raw gradient buffer into the push
// delta [ -lr*g/sqrt(G+g^2+eps) | g^2 ] using the accumulator half of
// the pulled rows.
__global__ void k_adagrad_rows(const float* __restrict__ grad, const float* __restrict__ rows,
                               float* __restrict__ out, int64_t n_rows, int dvals, int gstride,
                               int rstride, float lr, float eps) {
  int64_t total = n_rows * dvals;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t r = i / dvals;
    int k = (int)(i % dvals);
    float g = grad[r * gstride + k];
    float G = rows[r * rstride + dvals + k];
    out[r * rstride + k] = -lr * g * __frsqrt_rn(G + g * g + eps);
    out[r * rstride + dvals + k] = g * g;
  }
}

void adagrad_rows_gpu(const float* grad, const float* rows, float* out, int64_t n_rows,
                      int dvals, int gstride, int rstride, float lr, float eps, void* stream) {
  int64_t total = n_rows * dvals;
  if (total < 1) return;
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 8192);
  hipLaunchKernelGGL(k_adagrad_rows, dim3(blocks), dim3(256), 0, (hipStream_t)stream, grad,
                     rows, out, n_rows, dvals, gstride, rstride, lr, eps);
}

}  // namespace adapm
