// CPU backend of the app kernels (kernels.h) — identical math to
// kernels_hip.hip, used by the no-GPU test tier and as the reference for
// GPU numerics tests (alongside plain-torch fp32 references in tests/).
#include <cmath>
#include <vector>

#include "kernels.h"

namespace adapm {

static inline float sigmoidf_(float x) { return 1.f / (1.f + std::exp(-x)); }
static inline float softplusf_(float x) { return x > 20.f ? x : std::log1p(std::exp(x)); }

void kge_complex_step_cpu(const float* s, const float* r, const float* o, const float* neg,
                          float* ds, float* dr, float* do_, float* dneg, float* loss, int B,
                          int N, int D, float lr, float eps) {
  const int dc = D >> 1;
  const int row = D << 1;
  std::vector<float> a_sre(dc), a_sim(dc), a_rre(dc), a_rim(dc), u_re(dc), u_im(dc);
  for (int b = 0; b < B; ++b) {
    const float* sb = s + (int64_t)b * row;
    const float* rb = r + (int64_t)b * row;
    for (int k = 0; k < dc; ++k) {
      u_re[k] = sb[k] * rb[k] - sb[dc + k] * rb[dc + k];
      u_im[k] = sb[dc + k] * rb[k] + sb[k] * rb[dc + k];
      a_sre[k] = a_sim[k] = a_rre[k] = a_rim[k] = 0.f;
    }
    float lsum = 0.f;
    for (int j = 0; j <= N; ++j) {
      const float* ob = (j == 0) ? o + (int64_t)b * row : neg + ((int64_t)b * N + (j - 1)) * row;
      float* dob = (j == 0) ? do_ + (int64_t)b * row : dneg + ((int64_t)b * N + (j - 1)) * row;
      float y = (j == 0) ? 1.f : -1.f;
      double psi = 0.0;
      for (int k = 0; k < dc; ++k) psi += u_re[k] * ob[k] + u_im[k] * ob[dc + k];
      float c = -y * sigmoidf_(-y * (float)psi);
      lsum += softplusf_(-y * (float)psi);
      for (int k = 0; k < dc; ++k) {
        float o_re = ob[k], o_im = ob[dc + k];
        a_sre[k] += c * (rb[k] * o_re + rb[dc + k] * o_im);
        a_sim[k] += c * (rb[k] * o_im - rb[dc + k] * o_re);
        a_rre[k] += c * (sb[k] * o_re + sb[dc + k] * o_im);
        a_rim[k] += c * (sb[k] * o_im - sb[dc + k] * o_re);
        float g_re = c * u_re[k], g_im = c * u_im[k];
        dob[k] = -lr * g_re / std::sqrt(ob[D + k] + g_re * g_re + eps);
        dob[dc + k] = -lr * g_im / std::sqrt(ob[D + dc + k] + g_im * g_im + eps);
        dob[D + k] = g_re * g_re;
        dob[D + dc + k] = g_im * g_im;
      }
    }
    float* dsb = ds + (int64_t)b * row;
    float* drb = dr + (int64_t)b * row;
    for (int k = 0; k < dc; ++k) {
      dsb[k] = -lr * a_sre[k] / std::sqrt(sb[D + k] + a_sre[k] * a_sre[k] + eps);
      dsb[dc + k] = -lr * a_sim[k] / std::sqrt(sb[D + dc + k] + a_sim[k] * a_sim[k] + eps);
      dsb[D + k] = a_sre[k] * a_sre[k];
      dsb[D + dc + k] = a_sim[k] * a_sim[k];
      drb[k] = -lr * a_rre[k] / std::sqrt(rb[D + k] + a_rre[k] * a_rre[k] + eps);
      drb[dc + k] = -lr * a_rim[k] / std::sqrt(rb[D + dc + k] + a_rim[k] * a_rim[k] + eps);
      drb[D + k] = a_rre[k] * a_rre[k];
      drb[D + dc + k] = a_rim[k] * a_rim[k];
    }
    loss[b] = lsum;
  }
}

void kge_complex_score_cpu(const float* s, const float* r, const float* cand, float* scores,
                           int B, int E, int D) {
  const int dc = D >> 1;
  const int row = D << 1;
  for (int b = 0; b < B; ++b) {
    const float* sb = s + (int64_t)b * row;
    const float* rb = r + (int64_t)b * row;
    for (int e = 0; e < E; ++e) {
      const float* ob = cand + (int64_t)e * row;
      double psi = 0.0;
      for (int k = 0; k < dc; ++k) {
        float ure = sb[k] * rb[k] - sb[dc + k] * rb[dc + k];
        float uim = sb[dc + k] * rb[k] + sb[k] * rb[dc + k];
        psi += ure * ob[k] + uim * ob[dc + k];
      }
      scores[(int64_t)b * E + e] = (float)psi;
    }
  }
}

void w2v_sgns_step_cpu(const float* ctr, const float* ctx, const float* neg, float* dctr,
                       float* dctx, float* dneg, float* loss, int B, int N, int D, float lr,
                       float eps) {
  const int row = D << 1;
  std::vector<float> a_c(D);
  for (int b = 0; b < B; ++b) {
    const float* cb = ctr + (int64_t)b * row;
    std::fill(a_c.begin(), a_c.end(), 0.f);
    float lsum = 0.f;
    for (int j = 0; j <= N; ++j) {
      const float* xb = (j == 0) ? ctx + (int64_t)b * row : neg + ((int64_t)b * N + (j - 1)) * row;
      float* dxb = (j == 0) ? dctx + (int64_t)b * row : dneg + ((int64_t)b * N + (j - 1)) * row;
      float y = (j == 0) ? 1.f : -1.f;
      double dot = 0.0;
      for (int k = 0; k < D; ++k) dot += cb[k] * xb[k];
      float g = -y * sigmoidf_(-y * (float)dot);
      lsum += softplusf_(-y * (float)dot);
      for (int k = 0; k < D; ++k) {
        a_c[k] += g * xb[k];
        float gx = g * cb[k];
        dxb[k] = -lr * gx / std::sqrt(xb[D + k] + gx * gx + eps);
        dxb[D + k] = gx * gx;
      }
    }
    float* dcb = dctr + (int64_t)b * row;
    for (int k = 0; k < D; ++k) {
      dcb[k] = -lr * a_c[k] / std::sqrt(cb[D + k] + a_c[k] * a_c[k] + eps);
      dcb[D + k] = a_c[k] * a_c[k];
    }
    loss[b] = lsum;
  }
}

void mf_update_step_cpu(const float* w, const float* h, const float* x, float* dw, float* dh,
                        float* loss, int B, int R, float lr, float lambda, float eps) {
  const int row = R << 1;
  for (int b = 0; b < B; ++b) {
    const float* wb = w + (int64_t)b * row;
    const float* hb = h + (int64_t)b * row;
    double pred = 0.0;
    for (int k = 0; k < R; ++k) pred += wb[k] * hb[k];
    float e = x[b] - (float)pred;
    loss[b] = e * e;
    float* dwb = dw + (int64_t)b * row;
    float* dhb = dh + (int64_t)b * row;
    for (int k = 0; k < R; ++k) {
      float gw = -2.f * e * hb[k] + 2.f * lambda * wb[k];
      float gh = -2.f * e * wb[k] + 2.f * lambda * hb[k];
      dwb[k] = -lr * gw / std::sqrt(wb[R + k] + gw * gw + eps);
      dwb[R + k] = gw * gw;
      dhb[k] = -lr * gh / std::sqrt(hb[R + k] + gh * gh + eps);
      dhb[R + k] = gh * gh;
    }
  }
}

// Fused slab-direct steps, offsets mode (CPU tier of the world>1 fused
// path: offs_* carry float offsets into the slab, resolved by the host
// metadata pass; math mirrors the GPU fused kernels — object rows are
// updated immediately per negative, s/r accumulated then applied).
void kge_complex_step_fused_offs_cpu(float* slab, const int64_t* offs_s, const int64_t* offs_r,
                                     const int64_t* offs_o, const int64_t* offs_neg, float* loss,
                                     int B, int N, int D, float lr, float eps) {
  const int dc = D >> 1;
  std::vector<float> a_sre(dc), a_sim(dc), a_rre(dc), a_rim(dc), u_re(dc), u_im(dc);
  for (int b = 0; b < B; ++b) {
    float* sb = slab + offs_s[b];
    float* rb = slab + offs_r[b];
    for (int k = 0; k < dc; ++k) {
      u_re[k] = sb[k] * rb[k] - sb[dc + k] * rb[dc + k];
      u_im[k] = sb[dc + k] * rb[k] + sb[k] * rb[dc + k];
      a_sre[k] = a_sim[k] = a_rre[k] = a_rim[k] = 0.f;
    }
    float lsum = 0.f;
    for (int j = 0; j <= N; ++j) {
      float* ob = slab + ((j == 0) ? offs_o[b] : offs_neg[(int64_t)b * N + (j - 1)]);
      float y = (j == 0) ? 1.f : -1.f;
      float psi = 0.f;
      for (int k = 0; k < dc; ++k) psi += u_re[k] * ob[k] + u_im[k] * ob[dc + k];
      float c = -y * sigmoidf_(-y * psi);
      lsum += softplusf_(-y * psi);
      for (int k = 0; k < dc; ++k) {
        float o_re = ob[k], o_im = ob[dc + k];
        a_sre[k] += c * (rb[k] * o_re + rb[dc + k] * o_im);
        a_sim[k] += c * (rb[k] * o_im - rb[dc + k] * o_re);
        a_rre[k] += c * (sb[k] * o_re + sb[dc + k] * o_im);
        a_rim[k] += c * (sb[k] * o_im - sb[dc + k] * o_re);
        float g_re = c * u_re[k], g_im = c * u_im[k];
        float G_re = ob[D + k] + g_re * g_re;
        float G_im = ob[D + dc + k] + g_im * g_im;
        ob[k] += -lr * g_re / std::sqrt(G_re + eps);
        ob[dc + k] += -lr * g_im / std::sqrt(G_im + eps);
        ob[D + k] += g_re * g_re;
        ob[D + dc + k] += g_im * g_im;
      }
    }
    for (int k = 0; k < dc; ++k) {
      float Gsr = sb[D + k] + a_sre[k] * a_sre[k];
      float Gsi = sb[D + dc + k] + a_sim[k] * a_sim[k];
      sb[k] += -lr * a_sre[k] / std::sqrt(Gsr + eps);
      sb[dc + k] += -lr * a_sim[k] / std::sqrt(Gsi + eps);
      sb[D + k] += a_sre[k] * a_sre[k];
      sb[D + dc + k] += a_sim[k] * a_sim[k];
      float Grr = rb[D + k] + a_rre[k] * a_rre[k];
      float Gri = rb[D + dc + k] + a_rim[k] * a_rim[k];
      rb[k] += -lr * a_rre[k] / std::sqrt(Grr + eps);
      rb[dc + k] += -lr * a_rim[k] / std::sqrt(Gri + eps);
      rb[D + k] += a_rre[k] * a_rre[k];
      rb[D + dc + k] += a_rim[k] * a_rim[k];
    }
    loss[b] = lsum;
  }
}

void w2v_sgns_step_fused_offs_cpu(float* slab, const int64_t* offs_ctr, const int64_t* offs_ctx,
                                  const int64_t* offs_neg, float* loss, int B, int N, int D,
                                  float lr, float eps) {
  std::vector<float> a_c(D), c_emb(D);
  for (int b = 0; b < B; ++b) {
    float* cb = slab + offs_ctr[b];
    for (int k = 0; k < D; ++k) {
      c_emb[k] = cb[k];
      a_c[k] = 0.f;
    }
    float lsum = 0.f;
    for (int j = 0; j <= N; ++j) {
      float* xb = slab + ((j == 0) ? offs_ctx[b] : offs_neg[(int64_t)b * N + (j - 1)]);
      float y = (j == 0) ? 1.f : -1.f;
      float dot = 0.f;
      for (int k = 0; k < D; ++k) dot += c_emb[k] * xb[k];
      float g = -y * sigmoidf_(-y * dot);
      lsum += softplusf_(-y * dot);
      for (int k = 0; k < D; ++k) {
        float xv = xb[k];
        a_c[k] += g * xv;
        float gx = g * c_emb[k];
        float G = xb[D + k] + gx * gx;
        xb[k] += -lr * gx / std::sqrt(G + eps);
        xb[D + k] += gx * gx;
      }
    }
    for (int k = 0; k < D; ++k) {
      float G = cb[D + k] + a_c[k] * a_c[k];
      cb[k] += -lr * a_c[k] / std::sqrt(G + eps);
      cb[D + k] += a_c[k] * a_c[k];
    }
    loss[b] = lsum;
  }
}

void mf_update_step_fused_offs_cpu(float* slab, const int64_t* offs_w, const int64_t* offs_h,
                                   const float* x, float* loss, int B, int R, float lr,
                                   float lambda, float eps) {
  for (int b = 0; b < B; ++b) {
    float* wb = slab + offs_w[b];
    float* hb = slab + offs_h[b];
    float pred = 0.f;
    for (int k = 0; k < R; ++k) pred += wb[k] * hb[k];
    float e = x[b] - pred;
    loss[b] = e * e;
    for (int k = 0; k < R; ++k) {
      float wv = wb[k], hv = hb[k];
      float gw = -2.f * e * hv + 2.f * lambda * wv;
      float gh = -2.f * e * wv + 2.f * lambda * hv;
      float Gw = wb[R + k] + gw * gw;
      float Gh = hb[R + k] + gh * gh;
      wb[k] += -lr * gw / std::sqrt(Gw + eps);
      wb[R + k] += gw * gw;
      hb[k] += -lr * gh / std::sqrt(Gh + eps);
      hb[R + k] += gh * gh;
    }
  }
}

void mf_loss_cpu(const float* w, const float* h, const float* x, float* out2, int B, int R,
                 float lambda) {
  const int row = R << 1;
  double se = 0, reg = 0;
  for (int b = 0; b < B; ++b) {
    const float* wb = w + (int64_t)b * row;
    const float* hb = h + (int64_t)b * row;
    double pred = 0, nrm = 0;
    for (int k = 0; k < R; ++k) {
      pred += wb[k] * hb[k];
      nrm += wb[k] * wb[k] + hb[k] * hb[k];
    }
    double e = x[b] - pred;
    se += e * e;
    reg += lambda * nrm;
  }
  out2[0] += (float)se;
  out2[1] += (float)reg;
}

static inline uint64_t pcg_hash64_c(uint64_t x) {
  x ^= x >> 33; x *= 0xff51afd7ed558ccdULL;
  x ^= x >> 33; x *= 0xc4ceb9fe1a85ec53ULL;
  x ^= x >> 33;
  return x;
}

void alias_draw_cpu(const float* prob, const int32_t* alias, int64_t n, uint64_t seed,
                    int64_t N, int64_t* out) {
  for (int64_t i = 0; i < N; ++i) {
    uint64_t h = pcg_hash64_c(seed ^ (uint64_t)i * 0x9e3779b97f4a7c15ULL);
    int64_t slot = (int64_t)(h % (uint64_t)n);
    float u = (float)((h >> 40) & 0xffffff) * (1.0f / 16777216.0f);
    out[i] = (u < prob[slot]) ? slot : (int64_t)alias[slot];
  }
}

}  // namespace adapm

namespace adapm {

void rescal_step_cpu(const float* s, const float* r, const float* o, const float* neg,
                     float* ds, float* drl, float* do_, float* dneg, float* loss, int B, int N,
                     int D, float lr, float eps) {
  const int erow = 2 * D;
  const int64_t rrow = 2LL * D * D;
  std::vector<float> u(D), w(D), t(D);
  for (int b = 0; b < B; ++b) {
    const float* sb = s + (int64_t)b * erow;
    const float* rb = r + (int64_t)b * rrow;
    // u = R^T e_s
    for (int j = 0; j < D; ++j) {
      double a = 0;
      for (int i = 0; i < D; ++i) a += rb[(int64_t)i * D + j] * sb[i];
      u[j] = (float)a;
    }
    std::fill(w.begin(), w.end(), 0.f);
    float lsum = 0.f;
    for (int j = 0; j <= N; ++j) {
      const float* ob = (j == 0) ? o + (int64_t)b * erow : neg + ((int64_t)b * N + j - 1) * erow;
      float* dob = (j == 0) ? do_ + (int64_t)b * erow : dneg + ((int64_t)b * N + j - 1) * erow;
      float y = (j == 0) ? 1.f : -1.f;
      double dot = 0;
      for (int k = 0; k < D; ++k) dot += u[k] * ob[k];
      float c = -y * sigmoidf_(-y * (float)dot);
      lsum += softplusf_(-y * (float)dot);
      for (int k = 0; k < D; ++k) {
        w[k] += c * ob[k];
        float g = c * u[k];
        dob[k] = -lr * g / std::sqrt(ob[D + k] + g * g + eps);
        dob[D + k] = g * g;
      }
    }
    loss[b] = lsum;
    // de_s = R w
    float* dsb = ds + (int64_t)b * erow;
    for (int i = 0; i < D; ++i) {
      double a = 0;
      for (int k = 0; k < D; ++k) a += rb[(int64_t)i * D + k] * w[k];
      t[i] = (float)a;
      float g = t[i];
      dsb[i] = -lr * g / std::sqrt(sb[D + i] + g * g + eps);
      dsb[D + i] = g * g;
    }
    // dR = e_s w^T
    float* drb = drl + (int64_t)b * rrow;
    for (int i = 0; i < D; ++i) {
      for (int k = 0; k < D; ++k) {
        float g = sb[i] * w[k];
        int64_t idx = (int64_t)i * D + k;
        drb[idx] = -lr * g / std::sqrt(rb[(int64_t)D * D + idx] + g * g + eps);
        drb[(int64_t)D * D + idx] = g * g;
      }
    }
  }
}

}  // namespace adapm
