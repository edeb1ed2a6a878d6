// App compute kernels (the reference's per-datapoint scalar loops become
// batched device kernels — SURVEY.md §2.5 inventory):
//   - KGE ComplEx score/grad + fused AdaGrad (reference
//     knowledge_graph_embeddings.cc:832-858 score/grad, 415-435 AdaGrad)
//   - word2vec SGNS step (reference word2vec.cc:679-745)
//   - MF NZSL+L2 update (reference apps/mf/update.h:32-69)
//
// Value layout convention (same as the reference apps): every key's value
// is [embedding(D) | adagrad_accum(D)], so AdaGrad state rides the same
// Push/Pull path. Kernels consume pulled rows and produce PUSH-ready
// additive deltas: [delta_emb(D) | grad^2(D)].
#pragma once
#include <cstdint>

namespace adapm {

// ComplEx training step over B positives with N o-side negatives each.
//  s, r, o:   [B][2D]   pulled rows (emb | accum)
//  neg:       [B*N][2D] pulled negative-entity rows
//  ds,dr,do_,dneg: same shapes, outputs (push deltas)
//  loss:      [B] logistic loss of each positive (+ its negatives)
// D = embedding dim (even: first half real, second half imaginary).
void kge_complex_step_gpu(const float* s, const float* r, const float* o, const float* neg,
                          float* ds, float* dr, float* do_, float* dneg, float* loss,
                          int B, int N, int D, float lr, float eps, void* stream);
void kge_complex_step_cpu(const float* s, const float* r, const float* o, const float* neg,
                          float* ds, float* dr, float* do_, float* dneg, float* loss,
                          int B, int N, int D, float lr, float eps);

// FUSED ComplEx step: reads entity/relation rows DIRECTLY from the slab
// and atomically accumulates the AdaGrad-transformed deltas back — no
// intermediate pull/push buffers (the classic path pays a gather write +
// kernel read + delta write + scatter read; this saves all four).
// Duplicate keys within the batch see hogwild-style concurrent updates
// (async-PS semantics). keys_* are DEVICE int64 pointers.
// Addressing modes (row_at): world >= 1 — identity layout, key k at
// (k/world)*plen (single-rank fast path, zero host work); world == 0 —
// keys_* carry precomputed float OFFSETS into the slab (the world>1 /
// relocated-layout path: the host pass resolves offsets and compacts
// all-local samples, remote ones take the classic path).
void kge_complex_step_fused_gpu(float* slab, const int64_t* keys_s, const int64_t* keys_r,
                                const int64_t* keys_o, const int64_t* keys_neg, float* loss,
                                int B, int N, int D, int32_t plen, int world, int rank,
                                float lr, float eps, void* stream);

// CPU tier of the offsets-mode fused steps (exercises the world>1 fused
// protocol path in the gloo multi-process tests; serial, in-place).
void kge_complex_step_fused_offs_cpu(float* slab, const int64_t* offs_s, const int64_t* offs_r,
                                     const int64_t* offs_o, const int64_t* offs_neg, float* loss,
                                     int B, int N, int D, float lr, float eps);
void w2v_sgns_step_fused_offs_cpu(float* slab, const int64_t* offs_ctr, const int64_t* offs_ctx,
                                  const int64_t* offs_neg, float* loss, int B, int N, int D,
                                  float lr, float eps);
void mf_update_step_fused_offs_cpu(float* slab, const int64_t* offs_w, const int64_t* offs_h,
                                   const float* x, float* loss, int B, int R, float lr,
                                   float lambda, float eps);

// FUSED SGNS step (same contract as kge_complex_step_fused_gpu: slab-
// direct reads + atomic AdaGrad writes, hogwild on duplicates).
void w2v_sgns_step_fused_gpu(float* slab, const int64_t* keys_ctr, const int64_t* keys_ctx,
                             const int64_t* keys_neg, float* loss, int B, int N, int D,
                             int32_t plen, int world, float lr, float eps, void* stream);

// FUSED MF step (slab-direct NZSL+L2; same contract as above).
void mf_update_step_fused_gpu(float* slab, const int64_t* keys_w, const int64_t* keys_h,
                              const float* x, float* loss, int B, int R, int32_t plen,
                              int world, float lr, float lambda, float eps, void* stream);

// ComplEx scoring only (evaluation): score[b][e] = psi(s_b, r_b, cand_e)
//  cand: [E][2D] candidate entity rows; scores: [B][E]
void kge_complex_score_gpu(const float* s, const float* r, const float* cand, float* scores,
                           int B, int E, int D, void* stream);
void kge_complex_score_cpu(const float* s, const float* r, const float* cand, float* scores,
                           int B, int E, int D);

// RESCAL training step (reference knowledge_graph_embeddings.cc:895-922):
// score = e_s^T R e_o, R is D x D. Entity rows [emb(D) | accum(D)],
// relation rows [R(D*D) | accum(D*D)]. Math uses the factorization
// u = R^T e_s (once), w = sum_o c_o e_o, de_s = R w, dR = e_s w^T — so a
// triple with N negatives costs ~3 D^2 + N D instead of N D^2.
void rescal_step_gpu(const float* s, const float* r, const float* o, const float* neg,
                     float* ds, float* drl, float* do_, float* dneg, float* loss, int B, int N,
                     int D, float lr, float eps, void* stream);
void rescal_step_cpu(const float* s, const float* r, const float* o, const float* neg,
                     float* ds, float* drl, float* do_, float* dneg, float* loss, int B, int N,
                     int D, float lr, float eps);

// word2vec SGNS step: per center/context pair with N negatives.
//  ctr:  [B][2D] center (syn0) rows;  ctx: [B][2D] context (syn1) rows
//  neg:  [B*N][2D] negative (syn1) rows
//  outputs: push deltas, same shapes; loss [B]
void w2v_sgns_step_gpu(const float* ctr, const float* ctx, const float* neg, float* dctr,
                       float* dctx, float* dneg, float* loss, int B, int N, int D, float lr,
                       float eps, void* stream);
void w2v_sgns_step_cpu(const float* ctr, const float* ctx, const float* neg, float* dctr,
                       float* dctx, float* dneg, float* loss, int B, int N, int D, float lr,
                       float eps);

// Matrix-factorization NZSL step: per nonzero (i, j, x):
//  w: [B][2R] row-factor rows, h: [B][2R] col-factor rows (R = rank)
//  outputs dw, dh (push deltas), loss [B] squared error
void mf_update_step_gpu(const float* w, const float* h, const float* x, float* dw, float* dh,
                        float* loss, int B, int R, float lr, float lambda, float eps,
                        void* stream);
void mf_update_step_cpu(const float* w, const float* h, const float* x, float* dw, float* dh,
                        float* loss, int B, int R, float lr, float lambda, float eps);

// MF NZSL+L2 loss reduction over B nonzeros (reference apps/mf/loss.h:
// 49-120: sum (x - w.h)^2 + lambda*(|w|^2 + |h|^2) per nonzero):
// out[0] += squared-error sum, out[1] += regularizer sum.
void mf_loss_gpu(const float* w, const float* h, const float* x, float* out2, int B, int R,
                 float lambda, void* stream);
void mf_loss_cpu(const float* w, const float* h, const float* x, float* out2, int B, int R,
                 float lambda);

// Alias-table sampling draw (negative sampling; reference unigram table,
// word2vec.cc:125-146 — rebuilt as an O(1)-per-draw alias table):
//  prob[n] f32, alias[n] i32 built host-side; out[N] int64 drawn keys.
//  Counter-based RNG (PCG hash of (seed, index)): reproducible, stateless.
void alias_draw_gpu(const float* prob, const int32_t* alias, int64_t n, uint64_t seed,
                    int64_t N, int64_t* out, void* stream);
void alias_draw_cpu(const float* prob, const int32_t* alias, int64_t n, uint64_t seed,
                    int64_t N, int64_t* out);

}  // namespace adapm

namespace adapm {

// ---- MFMA (matrix-core) kernels, mfma_hip.hip (gfx950
// v_mfma_f32_16x16x4_f32: exact f32 at the vector rate, ~2.4-2.8x a
// VALU f32 GEMM). GPU-only: the CPU tier uses the scalar references.

// Full-entity ComplEx eval scoring as a GEMM: scores[B][E] = Q @ Cand^T
// where Q (built by an elementwise prologue into qbuf[B*D]) is the
// (s o r) query and Cand rows are [emb(D)|accum(D)].
void kge_complex_score_mfma_gpu(const float* s, const float* r, const float* cand,
                                float* scores, float* qbuf, int B, int E, int D, void* stream);

// Batched varying-M grouped GEMM (triples sorted by relation; group g =
// rows [starts[g], starts[g+1]), relation matrix g at Rm + g*rstride):
//   mode 0: Out[row] = S_g @ R_g        (U = R^T e_s per row)
//   mode 1: Out[row] = W_g @ R_g^T      (de_s raw)
//   mode 2: Out[g]   = S_g^T @ W_g      (dR raw, [G][D*D])
void mfma_grouped_gemm_gpu(const float* S, const float* W, const float* R, float* out,
                           const int* starts, int G, int D, int sstride, int wstride,
                           int64_t rstride, int out_rstride, int64_t ostride, int mode,
                           int total_tiles, void* stream);

// per-triple middle phase of the grouped RESCAL step (scores, object
// grads with fused AdaGrad, W accumulation) given precomputed U
void rescal_mid_gpu(const float* U, const float* o, const float* neg, float* do_, float* dneg,
                    float* Wm, float* loss, int B, int N, int D, float lr, float eps,
                    void* stream);

// AdaGrad epilogue: out rows = [ -lr*g/sqrt(G+g^2+eps) | g^2 ] from a
// raw-gradient buffer and the accumulator half of the pulled rows
void adagrad_rows_gpu(const float* grad, const float* rows, float* out, int64_t n_rows,
                      int dvals, int gstride, int rstride, float lr, float eps, void* stream);

}  // namespace adapm
