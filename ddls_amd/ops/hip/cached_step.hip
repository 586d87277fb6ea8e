// Fully-fused PPO minibatch step for the cached-models SGD mode (gfx950).
//
// The tuned PAC-ML policy is ~35k params and a minibatch is 128 samples over
// M <= 16 distinct workload-model graphs (~200 nodes total): the whole
// fwd+bwd is ~10 MFLOP.  Run as torch autograd it is ~146 dispatches per
// replay (42% zero-fill / grad-accumulate glue — profiles/
// kernel_stats_r02_cachedsgd.csv); here it is TWO kernels:
//
//   cached_step_fwd : GNN (2 MeanPool rounds) over the M static model
//                     graphs -> per-model mean pooling -> per-sample head
//                     (LN+graph MLP ++ pooled emb -> 2 FC branches + mask)
//                     -> fused PPO loss (stats accumulated on device)
//   cached_step_bwd : the ENTIRE analytic backward, weight grads written
//                     DIRECTLY into the flat grad buffer (no AccumulateGrad,
//                     no zero-fills) — deterministic (no atomics), so
//                     hipGraph replays are bit-stable.
//
// Exact-by-linearity gradient aggregation: per-sample dL/d emb reaches the
// (shared) per-model GNN summed over the minibatch — the same contraction
// index_select's backward performs, done here in one deterministic pass.
//
// Layer semantics mirror models/gnn.py (MeanPoolLayer / GNNPolicy): each
// module is LayerNorm -> Linear -> ReLU (module_depth=1), message =
// concat(hn[src], he), self-message = concat(hn[v], 0), out =
// (sum msgs + self) / (deg+1), zero for nodes with no in-edges (DGL
// update_all semantics); graph_module is LN -> Linear (no activation);
// branches are Linear-ReLU-Linear; mask added as clamp(log(mask)).
// Loss matches ppo_loss.hip (clipped surrogate + kl + clipped vf +
// entropy).  LayerNorm eps = 1e-5 (torch default), biased variance.
//
// One 256-thread workgroup per kernel: every phase is a flat-index loop
// with __syncthreads between phases; all row activations / rstd / xhat
// needed by the backward are stored in global scratch.
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <vector>

#define NTHREADS 256
#define LN_EPS 1e-5f

// ---- tensor-list indices (mirror rl/graph_step.py fused mode) ----
enum {
  // static graph
  C_Z0 = 0, C_E, C_SRC, C_DST, C_ORDER, C_INDPTR, C_SRC_ORDER, C_SRC_INDPTR,
  C_NODE_MODEL, C_MODEL_NPTR,
  // staged inputs
  C_MODEL_IDS, C_GF, C_MASK, C_ACTIONS, C_OLD_LOGP, C_ADV, C_VTARG, C_KL,
  // weights
  C_FLAT_P, C_FLAT_G, C_OFFS,
  // fwd activation scratch
  C_HN1, C_HE1, C_RE1, C_RS1, C_H1,
  C_HN2, C_HE2, C_RE2, C_RS2, C_H2, C_POOLED,
  C_XH_Z1, C_XH_E1, C_XH_M1E, C_XH_M1S, C_RST_Z1, C_RST_E1, C_RST_M1E,
  C_RST_M1S,
  C_XH_H2, C_XH_E2, C_XH_M2E, C_XH_M2S, C_RST_H2, C_RST_E2, C_RST_M2E,
  C_RST_M2S,
  C_XH34, C_FINAL, C_H1P, C_H1V, C_VALUES, C_P, C_LP, C_COEF, C_HENT,
  C_STATS,
  // bwd scratch
  C_GLOGITS, C_GVALUE, C_GH1P, C_GH1V, C_GFINAL, C_GU34,
  C_GPOOL, C_GH2, C_GME2, C_GMS2, C_GHN2, C_GHE2, C_GH1B,
  C_GME1, C_GMS1, C_GHN1, C_GHE1,
  C_GPN2, C_GPE2, C_GPN1, C_GPE1,
  C_GPRE1E, C_GPRE1S,
  C_NT
};

// weight slots within OFFS (order fixed; mirror _fused_offsets in Python)
enum {
  W_LN_N1_W = 0, W_LN_N1_B, W_N1_W, W_N1_B,
  W_LN_E1_W, W_LN_E1_B, W_E1_W, W_E1_B,
  W_LN_R1_W, W_LN_R1_B, W_R1_W, W_R1_B,
  W_LN_N2_W, W_LN_N2_B, W_N2_W, W_N2_B,
  W_LN_E2_W, W_LN_E2_B, W_E2_W, W_E2_B,
  W_LN_R2_W, W_LN_R2_B, W_R2_W, W_R2_B,
  W_LN_G_W, W_LN_G_B, W_G_W, W_G_B,
  W_P1_W, W_P1_B, W_P2_W, W_P2_B,
  W_V1_W, W_V1_B, W_V2_W, W_V2_B,
  W_COUNT
};

struct CachedDims {
  int N, E, M, B, A;
  int F0, FE, H, MSG, HID, OUT, GFin, GEMB, FIN, FC;
  float clip, vf_clip, vf_coef, ent_coef;
};

struct CachedPtrs {
  const float *z0, *e, *gf, *mask, *old_logp, *adv, *vtarg, *kl;
  const long *src, *dst, *order, *indptr, *src_order, *src_indptr;
  const long *node_model, *model_nptr, *model_ids, *actions;
  const float *flat_p;
  float *flat_g;
  const long *offs;
  float *hn1, *he1, *re1, *rs1, *h1;
  float *hn2, *he2, *re2, *rs2, *h2, *pooled;
  float *xh_z1, *xh_e1, *xh_m1e, *xh_m1s, *rst_z1, *rst_e1, *rst_m1e,
        *rst_m1s;
  float *xh_h2, *xh_e2, *xh_m2e, *xh_m2s, *rst_h2, *rst_e2, *rst_m2e,
        *rst_m2s;
  float *xh34, *fin, *h1p, *h1v, *values, *p, *lp, *coef, *hent, *stats;
  float *glogits, *gvalue, *gh1p, *gh1v, *gfinal, *gu34;
  float *gpool, *gh2, *gme2, *gms2, *ghn2, *ghe2, *gh1b;
  float *gme1, *gms1, *ghn1, *ghe1;
  float *gpn2, *gpe2, *gpn1, *gpe1;
  float *gpre1_e, *gpre1_s;
};

#define WP(slot) (P.flat_p + P.offs[slot])
#define WG(slot) (P.flat_g + P.offs[slot])

// LayerNorm + store xhat and rstd for the backward.  One thread per row.
__device__ void ln_row(const float* x, int D, float* xhat, float* rstd_out) {
  float mu = 0.f;
  for (int i = 0; i < D; ++i) mu += x[i];
  mu /= D;
  float var = 0.f;
  for (int i = 0; i < D; ++i) { float d = x[i] - mu; var += d * d; }
  var /= D;
  float rstd = rsqrtf(var + LN_EPS);
  for (int i = 0; i < D; ++i) xhat[i] = (x[i] - mu) * rstd;
  *rstd_out = rstd;
}

// grad of LN wrt input, given upstream grad on the AFFINE output.
// gin = rstd * (ghat - mean(ghat) - xhat * mean(ghat*xhat)),  ghat = go*gamma
__device__ void ln_bwd_row(const float* go, const float* gamma,
                           const float* xhat, float rstd, int D, float* gin) {
  float m1 = 0.f, m2 = 0.f;
  for (int i = 0; i < D; ++i) {
    float gh = go[i] * gamma[i];
    m1 += gh;
    m2 += gh * xhat[i];
  }
  m1 /= D;
  m2 /= D;
  for (int i = 0; i < D; ++i) {
    float gh = go[i] * gamma[i];
    gin[i] = rstd * (gh - m1 - xhat[i] * m2);
  }
}

// one MeanPool "row MLP": out = relu(Lin(LNaffine(x)));  LN xhat/rstd saved
__device__ void row_mlp_fwd(const CachedPtrs& P, const float* x, int Din,
                            int Dout, int w_slot, float* xhat, float* rstd,
                            float* out) {
  ln_row(x, Din, xhat, rstd);
  const float* gam = WP(w_slot);        // LN weight slot
  const float* bet = WP(w_slot + 1);
  const float* W = WP(w_slot + 2);      // Linear weight [Dout, Din]
  const float* b = WP(w_slot + 3);
  for (int o = 0; o < Dout; ++o) {
    float acc = b[o];
    for (int i = 0; i < Din; ++i)
      acc += W[o * Din + i] * (xhat[i] * gam[i] + bet[i]);
    out[o] = acc > 0.f ? acc : 0.f;
  }
}

// backward of one row MLP row: given gout (wrt relu output) and stored
// (out, xhat, rstd): gpre = gout*(out>0); optional gin; gpre stored for the
// weight-grad pass.
__device__ void row_mlp_bwd_row(const CachedPtrs& P, const float* gout,
                                const float* out, const float* xhat,
                                float rstd, int Din, int Dout, int w_slot,
                                float* gpre_store, float* gin /*or null*/) {
  float gpre[64];
  for (int o = 0; o < Dout; ++o)
    gpre[o] = out[o] > 0.f ? gout[o] : 0.f;
  for (int o = 0; o < Dout; ++o) gpre_store[o] = gpre[o];
  if (gin != nullptr) {
    const float* W = WP(w_slot + 2);
    const float* gam = WP(w_slot);
    float gu[64];
    for (int i = 0; i < Din; ++i) {
      float acc = 0.f;
      for (int o = 0; o < Dout; ++o) acc += W[o * Din + i] * gpre[o];
      gu[i] = acc;
    }
    ln_bwd_row(gu, gam, xhat, rstd, Din, gin);
  }
}

// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(NTHREADS)
cached_step_fwd_kernel(CachedPtrs P, CachedDims D) {
  const int tid = threadIdx.x;
  const int NT = NTHREADS;

  // phase 1: round-1 node/edge MLPs (one thread per row)
  for (int v = tid; v < D.N; v += NT)
    row_mlp_fwd(P, P.z0 + (long)v * D.F0, D.F0, D.H, W_LN_N1_W,
                P.xh_z1 + (long)v * D.F0, P.rst_z1 + v,
                P.hn1 + (long)v * D.H);
  for (int k = tid; k < D.E; k += NT)
    row_mlp_fwd(P, P.e + (long)k * D.FE, D.FE, D.H, W_LN_E1_W,
                P.xh_e1 + (long)k * D.FE, P.rst_e1 + k,
                P.he1 + (long)k * D.H);
  __syncthreads();

  // phase 2: round-1 reduce MLP on edge + self messages
  for (int k = tid; k < D.E; k += NT) {
    const long s = P.src[k];
    float msg[64];
    for (int i = 0; i < D.H; ++i) msg[i] = P.hn1[s * D.H + i];
    for (int i = 0; i < D.H; ++i) msg[D.H + i] = P.he1[(long)k * D.H + i];
    row_mlp_fwd(P, msg, D.MSG, D.HID, W_LN_R1_W,
                P.xh_m1e + (long)k * D.MSG, P.rst_m1e + k,
                P.re1 + (long)k * D.HID);
  }
  for (int v = tid; v < D.N; v += NT) {
    float msg[64];
    for (int i = 0; i < D.H; ++i) msg[i] = P.hn1[(long)v * D.H + i];
    for (int i = 0; i < D.H; ++i) msg[D.H + i] = 0.f;
    row_mlp_fwd(P, msg, D.MSG, D.HID, W_LN_R1_W,
                P.xh_m1s + (long)v * D.MSG, P.rst_m1s + v,
                P.rs1 + (long)v * D.HID);
  }
  __syncthreads();

  // phase 3: round-1 combine (CSR by dst): h1 = (sum + self)/(deg+1), or 0
  for (long u = tid; u < (long)D.N * D.HID; u += NT) {
    int v = (int)(u / D.HID), i = (int)(u % D.HID);
    long lo = P.indptr[v], hi = P.indptr[v + 1];
    float out = 0.f;
    if (hi > lo) {
      float acc = P.rs1[(long)v * D.HID + i];
      for (long q = lo; q < hi; ++q)
        acc += P.re1[P.order[q] * D.HID + i];
      out = acc / (float)(hi - lo + 1);
    }
    P.h1[(long)v * D.HID + i] = out;
  }
  __syncthreads();

  // phase 4: round-2 node/edge MLPs
  for (int v = tid; v < D.N; v += NT)
    row_mlp_fwd(P, P.h1 + (long)v * D.HID, D.HID, D.H, W_LN_N2_W,
                P.xh_h2 + (long)v * D.HID, P.rst_h2 + v,
                P.hn2 + (long)v * D.H);
  for (int k = tid; k < D.E; k += NT)
    row_mlp_fwd(P, P.e + (long)k * D.FE, D.FE, D.H, W_LN_E2_W,
                P.xh_e2 + (long)k * D.FE, P.rst_e2 + k,
                P.he2 + (long)k * D.H);
  __syncthreads();

  // phase 5: round-2 reduce MLP
  for (int k = tid; k < D.E; k += NT) {
    const long s = P.src[k];
    float msg[64];
    for (int i = 0; i < D.H; ++i) msg[i] = P.hn2[s * D.H + i];
    for (int i = 0; i < D.H; ++i) msg[D.H + i] = P.he2[(long)k * D.H + i];
    row_mlp_fwd(P, msg, D.MSG, D.OUT, W_LN_R2_W,
                P.xh_m2e + (long)k * D.MSG, P.rst_m2e + k,
                P.re2 + (long)k * D.OUT);
  }
  for (int v = tid; v < D.N; v += NT) {
    float msg[64];
    for (int i = 0; i < D.H; ++i) msg[i] = P.hn2[(long)v * D.H + i];
    for (int i = 0; i < D.H; ++i) msg[D.H + i] = 0.f;
    row_mlp_fwd(P, msg, D.MSG, D.OUT, W_LN_R2_W,
                P.xh_m2s + (long)v * D.MSG, P.rst_m2s + v,
                P.rs2 + (long)v * D.OUT);
  }
  __syncthreads();

  // phase 6: round-2 combine -> h2
  for (long u = tid; u < (long)D.N * D.OUT; u += NT) {
    int v = (int)(u / D.OUT), i = (int)(u % D.OUT);
    long lo = P.indptr[v], hi = P.indptr[v + 1];
    float out = 0.f;
    if (hi > lo) {
      float acc = P.rs2[(long)v * D.OUT + i];
      for (long q = lo; q < hi; ++q)
        acc += P.re2[P.order[q] * D.OUT + i];
      out = acc / (float)(hi - lo + 1);
    }
    P.h2[(long)v * D.OUT + i] = out;
  }
  __syncthreads();

  // phase 7: per-model mean pooling
  for (long u = tid; u < (long)D.M * D.OUT; u += NT) {
    int m = (int)(u / D.OUT), i = (int)(u % D.OUT);
    long lo = P.model_nptr[m], hi = P.model_nptr[m + 1];
    float acc = 0.f;
    for (long v = lo; v < hi; ++v) acc += P.h2[v * D.OUT + i];
    P.pooled[(long)m * D.OUT + i] = acc / (float)(hi - lo);
  }
  __syncthreads();

  // phase 8: per-sample head: final = [pooled(model) ++ Lin(LN(gf))]
  for (int b = tid; b < D.B; b += NT) {
    const long mid = P.model_ids[b];
    float* fin = P.fin + (long)b * D.FIN;
    for (int i = 0; i < D.OUT; ++i) fin[i] = P.pooled[mid * D.OUT + i];
    float* xh = P.xh34 + (long)b * D.GFin;
    float rstd;
    ln_row(P.gf + (long)b * D.GFin, D.GFin, xh, &rstd);  // rstd not needed
    const float* gam = WP(W_LN_G_W);
    const float* bet = WP(W_LN_G_B);
    const float* Wg = WP(W_G_W);
    const float* bg = WP(W_G_B);
    for (int o = 0; o < D.GEMB; ++o) {
      float acc = bg[o];
      for (int i = 0; i < D.GFin; ++i)
        acc += Wg[o * D.GFin + i] * (xh[i] * gam[i] + bet[i]);
      fin[D.OUT + o] = acc;               // graph_module has NO activation
    }
  }
  __syncthreads();

  // phase 9: FC branches, unit-parallel: h1p/h1v
  for (long u = tid; u < (long)D.B * D.FC; u += NT) {
    int b = (int)(u / D.FC), j = (int)(u % D.FC);
    const float* fin = P.fin + (long)b * D.FIN;
    const float* W1p = WP(W_P1_W);
    const float* W1v = WP(W_V1_W);
    float ap = WP(W_P1_B)[j], av = WP(W_V1_B)[j];
    for (int i = 0; i < D.FIN; ++i) {
      ap += W1p[j * D.FIN + i] * fin[i];
      av += W1v[j * D.FIN + i] * fin[i];
    }
    P.h1p[u] = ap > 0.f ? ap : 0.f;
    P.h1v[u] = av > 0.f ? av : 0.f;
  }
  __syncthreads();

  // phase 10: logits (+mask) and value; then per-sample loss rows
  for (long u = tid; u < (long)D.B * D.A; u += NT) {
    int b = (int)(u / D.A), a = (int)(u % D.A);
    const float* W2p = WP(W_P2_W);
    float acc = WP(W_P2_B)[a];
    const float* h = P.h1p + (long)b * D.FC;
    for (int j = 0; j < D.FC; ++j) acc += W2p[a * D.FC + j] * h[j];
    float mk = P.mask[(long)b * D.A + a];
    float lm = logf(mk);
    const float fmin = -3.402823466e+38f;
    if (!(lm > fmin)) lm = fmin;          // clamp(log(mask), min=f32 min)
    P.p[u] = acc + lm;                    // reuse p as raw masked logits
  }
  for (int b = tid; b < D.B; b += NT) {
    const float* W2v = WP(W_V2_W);
    float acc = WP(W_V2_B)[0];
    const float* h = P.h1v + (long)b * D.FC;
    for (int j = 0; j < D.FC; ++j) acc += W2v[j] * h[j];
    P.values[b] = acc;
  }
  __syncthreads();

  // phase 11: loss per sample (softmax + surrogate), block-accumulate stats
  __shared__ float acc_s[4];
  if (tid < 4) acc_s[tid] = 0.f;
  __syncthreads();
  for (int b = tid; b < D.B; b += NT) {
    float* row = P.p + (long)b * D.A;
    float mx = -3.0e38f;
    for (int a = 0; a < D.A; ++a) mx = fmaxf(mx, row[a]);
    float Z = 0.f;
    for (int a = 0; a < D.A; ++a) Z += __expf(row[a] - mx);
    float lse = mx + __logf(Z);
    float ent = 0.f;
    for (int a = 0; a < D.A; ++a) {
      float lp = row[a] - lse;
      float p = __expf(lp);
      P.lp[(long)b * D.A + a] = lp;
      ent -= p * lp;
    }
    for (int a = 0; a < D.A; ++a)
      P.p[(long)b * D.A + a] = __expf(P.lp[(long)b * D.A + a]);
    const long a = P.actions[b];
    const float logp_a = P.lp[(long)b * D.A + a];
    const float ratio = __expf(logp_a - P.old_logp[b]);
    const float A_b = P.adv[b];
    const float r_cl = fminf(fmaxf(ratio, 1.f - D.clip), 1.f + D.clip);
    const float surr1 = ratio * A_b, surr2 = r_cl * A_b;
    const float surr = fminf(surr1, surr2);
    float ds = ratio * A_b;
    if (surr2 < surr1 && (ratio < 1.f - D.clip || ratio > 1.f + D.clip))
      ds = 0.f;
    const float kl_b = P.old_logp[b] - logp_a;
    const float verr = P.values[b] - P.vtarg[b];
    const float vf_b = fminf(verr * verr, D.vf_clip);
    P.hent[b] = ent;
    P.coef[b] = -ds - P.kl[0];
    atomicAdd(&acc_s[0], -surr);
    atomicAdd(&acc_s[1], vf_b);
    atomicAdd(&acc_s[2], kl_b);
    atomicAdd(&acc_s[3], ent);
  }
  __syncthreads();
  if (tid == 0) {
    const float inv = 1.f / D.B;
    const float pl = acc_s[0] * inv;
    const float vf = acc_s[1] * inv;
    const float kl = acc_s[2] * inv;
    const float ent = acc_s[3] * inv;
    const float loss = pl + P.kl[0] * kl + D.vf_coef * vf
                       - D.ent_coef * ent;
    P.stats[0] += pl;
    P.stats[1] += vf;
    P.stats[2] += kl;
    P.stats[3] += ent;
    P.stats[4] += loss;
  }
}

// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(NTHREADS)
cached_step_bwd_kernel(CachedPtrs P, CachedDims D) {
  const int tid = threadIdx.x;
  const int NT = NTHREADS;

  // phase 0: loss backward rows -> glogits, gvalue  (upstream grad = 1)
  for (long u = tid; u < (long)D.B * D.A; u += NT) {
    int b = (int)(u / D.A), a = (int)(u % D.A);
    const long act = P.actions[b];
    const float c = P.coef[b] / D.B;
    const float pj = P.p[u];
    const float lpj = P.lp[u];
    const float delta = (a == (int)act) ? 1.f : 0.f;
    P.glogits[u] = c * (delta - pj)
                   + (D.ent_coef / D.B) * pj * (lpj + P.hent[b]);
  }
  for (int b = tid; b < D.B; b += NT) {
    const float verr = P.values[b] - P.vtarg[b];
    const float mk = (verr * verr <= D.vf_clip) ? 1.f : 0.f;
    P.gvalue[b] = D.vf_coef * 2.f * verr * mk / D.B;
  }
  __syncthreads();

  // phase 1: branch hidden grads
  for (long u = tid; u < (long)D.B * D.FC; u += NT) {
    int b = (int)(u / D.FC), j = (int)(u % D.FC);
    const float* W2p = WP(W_P2_W);
    float acc = 0.f;
    for (int a = 0; a < D.A; ++a)
      acc += W2p[a * D.FC + j] * P.glogits[(long)b * D.A + a];
    P.gh1p[u] = P.h1p[u] > 0.f ? acc : 0.f;
    const float* W2v = WP(W_V2_W);
    float av = W2v[j] * P.gvalue[b];
    P.gh1v[u] = P.h1v[u] > 0.f ? av : 0.f;
  }
  __syncthreads();

  // phase 2: gfinal
  for (long u = tid; u < (long)D.B * D.FIN; u += NT) {
    int b = (int)(u / D.FIN), i = (int)(u % D.FIN);
    const float* W1p = WP(W_P1_W);
    const float* W1v = WP(W_V1_W);
    float acc = 0.f;
    for (int j = 0; j < D.FC; ++j) {
      acc += W1p[j * D.FIN + i] * P.gh1p[(long)b * D.FC + j];
      acc += W1v[j * D.FIN + i] * P.gh1v[(long)b * D.FC + j];
    }
    P.gfinal[u] = acc;
  }
  __syncthreads();

  // phase 3: graph-module LN-output grads (gu34) + pooled grads
  for (long u = tid; u < (long)D.B * D.GFin; u += NT) {
    int b = (int)(u / D.GFin), i = (int)(u % D.GFin);
    const float* Wg = WP(W_G_W);
    float acc = 0.f;
    for (int o = 0; o < D.GEMB; ++o)
      acc += Wg[o * D.GFin + i] * P.gfinal[(long)b * D.FIN + D.OUT + o];
    P.gu34[u] = acc;
  }
  for (long u = tid; u < (long)D.M * D.OUT; u += NT) {
    int m = (int)(u / D.OUT), i = (int)(u % D.OUT);
    float acc = 0.f;
    for (int b = 0; b < D.B; ++b)
      if ((int)P.model_ids[b] == m)
        acc += P.gfinal[(long)b * D.FIN + i];
    P.gpool[u] = acc;
  }
  __syncthreads();

  // phase 4: gh2 (pool-mean broadcast)
  for (long u = tid; u < (long)D.N * D.OUT; u += NT) {
    int v = (int)(u / D.OUT), i = (int)(u % D.OUT);
    const long m = P.node_model[v];
    const float nn = (float)(P.model_nptr[m + 1] - P.model_nptr[m]);
    P.gh2[u] = P.gpool[m * D.OUT + i] / nn;
  }
  __syncthreads();

  // phase 5: round-2 message grads
  for (int k = tid; k < D.E; k += NT) {
    const int v = (int)P.dst[k];
    const long deg = P.indptr[v + 1] - P.indptr[v];
    const float f = 1.f / (float)(deg + 1);
    float gr[64];
    for (int i = 0; i < D.OUT; ++i)
      gr[i] = P.gh2[(long)v * D.OUT + i] * f;
    row_mlp_bwd_row(P, gr, P.re2 + (long)k * D.OUT,
                    P.xh_m2e + (long)k * D.MSG, P.rst_m2e[k], D.MSG, D.OUT,
                    W_LN_R2_W, P.gpe2 + (long)k * D.OUT,
                    P.gme2 + (long)k * D.MSG);
  }
  for (int v = tid; v < D.N; v += NT) {
    const long deg = P.indptr[v + 1] - P.indptr[v];
    float gr[64];
    if (deg > 0) {
      const float f = 1.f / (float)(deg + 1);
      for (int i = 0; i < D.OUT; ++i)
        gr[i] = P.gh2[(long)v * D.OUT + i] * f;
    } else {
      for (int i = 0; i < D.OUT; ++i) gr[i] = 0.f;   // zero-filled output
    }
    row_mlp_bwd_row(P, gr, P.rs2 + (long)v * D.OUT,
                    P.xh_m2s + (long)v * D.MSG, P.rst_m2s[v], D.MSG, D.OUT,
                    W_LN_R2_W, P.gpn2 + (long)v * D.OUT,
                    P.gms2 + (long)v * D.MSG);
  }
  __syncthreads();

  // phase 6: scatter message grads to hn2 / he2 (CSR by src for hn)
  for (long u = tid; u < (long)D.N * D.H; u += NT) {
    int v = (int)(u / D.H), i = (int)(u % D.H);
    long lo = P.src_indptr[v], hi = P.src_indptr[v + 1];
    float acc = P.gms2[(long)v * D.MSG + i];      // self-message front half
    for (long q = lo; q < hi; ++q)
      acc += P.gme2[P.src_order[q] * D.MSG + i];
    P.ghn2[u] = acc;
  }
  for (long u = tid; u < (long)D.E * D.H; u += NT) {
    int k = (int)(u / D.H), i = (int)(u % D.H);
    P.ghe2[u] = P.gme2[(long)k * D.MSG + D.H + i];
  }
  __syncthreads();

  // phase 7: round-2 node/edge row-MLP backward (node input = h1 grads)
  for (int v = tid; v < D.N; v += NT)
    row_mlp_bwd_row(P, P.ghn2 + (long)v * D.H, P.hn2 + (long)v * D.H,
                    P.xh_h2 + (long)v * D.HID, P.rst_h2[v], D.HID, D.H,
                    W_LN_N2_W, P.gpn1 + (long)v * D.H /*reuse as gpre_n2*/,
                    P.gh1b + (long)v * D.HID);
  for (int k = tid; k < D.E; k += NT)
    row_mlp_bwd_row(P, P.ghe2 + (long)k * D.H, P.he2 + (long)k * D.H,
                    P.xh_e2 + (long)k * D.FE, P.rst_e2[k], D.FE, D.H,
                    W_LN_E2_W, P.gpe1 + (long)k * D.H /*reuse as gpre_e2m*/,
                    nullptr);
  __syncthreads();

  // ---- weight grads for round 2 + head are computed in phase W below;
  // first finish round-1 data flow (needs gh1) ----

  // phase 8: round-1 message grads (upstream gh1b); round-1 needs BOTH
  // gpre rows (HID wide, for the weight-grad pass) and gmsg rows (MSG wide)
  float* gpre1_e = P.gpre1_e;   // [E, HID]
  float* gpre1_s = P.gpre1_s;   // [N, HID]
  for (int k = tid; k < D.E; k += NT) {
    const int v = (int)P.dst[k];
    const long deg = P.indptr[v + 1] - P.indptr[v];
    const float f = 1.f / (float)(deg + 1);
    float gr[64];
    for (int i = 0; i < D.HID; ++i)
      gr[i] = P.gh1b[(long)v * D.HID + i] * f;
    row_mlp_bwd_row(P, gr, P.re1 + (long)k * D.HID,
                    P.xh_m1e + (long)k * D.MSG, P.rst_m1e[k], D.MSG, D.HID,
                    W_LN_R1_W, gpre1_e + (long)k * D.HID,
                    P.gme1 + (long)k * D.MSG);
  }
  for (int v = tid; v < D.N; v += NT) {
    const long deg = P.indptr[v + 1] - P.indptr[v];
    float gr[64];
    if (deg > 0) {
      const float f = 1.f / (float)(deg + 1);
      for (int i = 0; i < D.HID; ++i)
        gr[i] = P.gh1b[(long)v * D.HID + i] * f;
    } else {
      for (int i = 0; i < D.HID; ++i) gr[i] = 0.f;
    }
    row_mlp_bwd_row(P, gr, P.rs1 + (long)v * D.HID,
                    P.xh_m1s + (long)v * D.MSG, P.rst_m1s[v], D.MSG, D.HID,
                    W_LN_R1_W, gpre1_s + (long)v * D.HID,
                    P.gms1 + (long)v * D.MSG);
  }
  __syncthreads();

  // phase 9: scatter to hn1/he1 grads, then node/edge row-MLP gpre rows
  for (long u = tid; u < (long)D.N * D.H; u += NT) {
    int v = (int)(u / D.H), i = (int)(u % D.H);
    long lo = P.src_indptr[v], hi = P.src_indptr[v + 1];
    float acc = P.gms1[(long)v * D.MSG + i];
    for (long q = lo; q < hi; ++q)
      acc += P.gme1[P.src_order[q] * D.MSG + i];
    P.ghn1[u] = acc;
  }
  for (long u = tid; u < (long)D.E * D.H; u += NT) {
    int k = (int)(u / D.H), i = (int)(u % D.H);
    P.ghe1[u] = P.gme1[(long)k * D.MSG + D.H + i];
  }
  __syncthreads();
  // gpre rows for node1/edge1 (inputs static: no input grads) — store into
  // ghn1/ghe1 in place (relu mask applied)
  for (long u = tid; u < (long)D.N * D.H; u += NT)
    P.ghn1[u] = P.hn1[u] > 0.f ? P.ghn1[u] : 0.f;
  for (long u = tid; u < (long)D.E * D.H; u += NT)
    P.ghe1[u] = P.he1[u] > 0.f ? P.ghe1[u] : 0.f;
  __syncthreads();

  // ---- phase W: ALL weight gradients, deterministic unit-per-thread ----
  // helper lambdas are spelled out per weight for clarity.

  // FC branch 2: W2p[A,FC], b2p[A]; W2v[1,FC], b2v[1]
  for (long u = tid; u < (long)D.A * D.FC; u += NT) {
    int a = (int)(u / D.FC), j = (int)(u % D.FC);
    float acc = 0.f;
    for (int b = 0; b < D.B; ++b)
      acc += P.glogits[(long)b * D.A + a] * P.h1p[(long)b * D.FC + j];
    WG(W_P2_W)[u] = acc;
  }
  for (int a = tid; a < D.A; a += NT) {
    float acc = 0.f;
    for (int b = 0; b < D.B; ++b) acc += P.glogits[(long)b * D.A + a];
    WG(W_P2_B)[a] = acc;
  }
  for (int j = tid; j < D.FC; j += NT) {
    float acc = 0.f, accb = 0.f;
    for (int b = 0; b < D.B; ++b) {
      acc += P.gvalue[b] * P.h1v[(long)b * D.FC + j];
      accb += P.gvalue[b];
    }
    WG(W_V2_W)[j] = acc;
    if (j == 0) WG(W_V2_B)[0] = accb;
  }
  // FC branch 1: W1p[FC,FIN], b1p[FC]; W1v, b1v
  for (long u = tid; u < (long)D.FC * D.FIN; u += NT) {
    int j = (int)(u / D.FIN), i = (int)(u % D.FIN);
    float ap = 0.f, av = 0.f;
    for (int b = 0; b < D.B; ++b) {
      const float fin = P.fin[(long)b * D.FIN + i];
      ap += P.gh1p[(long)b * D.FC + j] * fin;
      av += P.gh1v[(long)b * D.FC + j] * fin;
    }
    WG(W_P1_W)[u] = ap;
    WG(W_V1_W)[u] = av;
  }
  for (int j = tid; j < D.FC; j += NT) {
    float ap = 0.f, av = 0.f;
    for (int b = 0; b < D.B; ++b) {
      ap += P.gh1p[(long)b * D.FC + j];
      av += P.gh1v[(long)b * D.FC + j];
    }
    WG(W_P1_B)[j] = ap;
    WG(W_V1_B)[j] = av;
  }
  // graph module: Wg[GEMB,GFin] += ggemb x u34 (u = xhat*gam+bet), bg, LN
  {
    const float* gam = WP(W_LN_G_W);
    const float* bet = WP(W_LN_G_B);
    for (long u = tid; u < (long)D.GEMB * D.GFin; u += NT) {
      int o = (int)(u / D.GFin), i = (int)(u % D.GFin);
      float acc = 0.f;
      for (int b = 0; b < D.B; ++b) {
        const float xh = P.xh34[(long)b * D.GFin + i];
        acc += P.gfinal[(long)b * D.FIN + D.OUT + o]
               * (xh * gam[i] + bet[i]);
      }
      WG(W_G_W)[u] = acc;
    }
    for (int o = tid; o < D.GEMB; o += NT) {
      float acc = 0.f;
      for (int b = 0; b < D.B; ++b)
        acc += P.gfinal[(long)b * D.FIN + D.OUT + o];
      WG(W_G_B)[o] = acc;
    }
    for (int i = tid; i < D.GFin; i += NT) {
      float gw = 0.f, gb = 0.f;
      for (int b = 0; b < D.B; ++b) {
        const float gu = P.gu34[(long)b * D.GFin + i];
        gw += gu * P.xh34[(long)b * D.GFin + i];
        gb += gu;
      }
      WG(W_LN_G_W)[i] = gw;
      WG(W_LN_G_B)[i] = gb;
    }
  }
  __syncthreads();

  // reduce MLP 2: Wr2[OUT,MSG] over E edge rows + N self rows
  {
    const float* gam = WP(W_LN_R2_W);
    const float* bet = WP(W_LN_R2_B);
    for (long u = tid; u < (long)D.OUT * D.MSG; u += NT) {
      int o = (int)(u / D.MSG), i = (int)(u % D.MSG);
      float acc = 0.f;
      for (int k = 0; k < D.E; ++k)
        acc += P.gpe2[(long)k * D.OUT + o]
               * (P.xh_m2e[(long)k * D.MSG + i] * gam[i] + bet[i]);
      for (int v = 0; v < D.N; ++v)
        acc += P.gpn2[(long)v * D.OUT + o]
               * (P.xh_m2s[(long)v * D.MSG + i] * gam[i] + bet[i]);
      WG(W_R2_W)[u] = acc;
    }
    for (int o = tid; o < D.OUT; o += NT) {
      float acc = 0.f;
      for (int k = 0; k < D.E; ++k) acc += P.gpe2[(long)k * D.OUT + o];
      for (int v = 0; v < D.N; ++v) acc += P.gpn2[(long)v * D.OUT + o];
      WG(W_R2_B)[o] = acc;
    }
    const float* Wr = WP(W_R2_W);
    for (int i = tid; i < D.MSG; i += NT) {
      float gw = 0.f, gb = 0.f;
      for (int k = 0; k < D.E; ++k) {
        float gu = 0.f;
        for (int o = 0; o < D.OUT; ++o)
          gu += Wr[o * D.MSG + i] * P.gpe2[(long)k * D.OUT + o];
        gw += gu * P.xh_m2e[(long)k * D.MSG + i];
        gb += gu;
      }
      for (int v = 0; v < D.N; ++v) {
        float gu = 0.f;
        for (int o = 0; o < D.OUT; ++o)
          gu += Wr[o * D.MSG + i] * P.gpn2[(long)v * D.OUT + o];
        gw += gu * P.xh_m2s[(long)v * D.MSG + i];
        gb += gu;
      }
      WG(W_LN_R2_W)[i] = gw;
      WG(W_LN_R2_B)[i] = gb;
    }
  }
  // node module 2: Wn2[H,HID] over N rows (gpre in gpn1 buffer)
  {
    const float* gam = WP(W_LN_N2_W);
    const float* bet = WP(W_LN_N2_B);
    for (long u = tid; u < (long)D.H * D.HID; u += NT) {
      int o = (int)(u / D.HID), i = (int)(u % D.HID);
      float acc = 0.f;
      for (int v = 0; v < D.N; ++v)
        acc += P.gpn1[(long)v * D.H + o]
               * (P.xh_h2[(long)v * D.HID + i] * gam[i] + bet[i]);
      WG(W_N2_W)[u] = acc;
    }
    for (int o = tid; o < D.H; o += NT) {
      float acc = 0.f;
      for (int v = 0; v < D.N; ++v) acc += P.gpn1[(long)v * D.H + o];
      WG(W_N2_B)[o] = acc;
    }
    const float* Wn = WP(W_N2_W);
    for (int i = tid; i < D.HID; i += NT) {
      float gw = 0.f, gb = 0.f;
      for (int v = 0; v < D.N; ++v) {
        float gu = 0.f;
        for (int o = 0; o < D.H; ++o)
          gu += Wn[o * D.HID + i] * P.gpn1[(long)v * D.H + o];
        gw += gu * P.xh_h2[(long)v * D.HID + i];
        gb += gu;
      }
      WG(W_LN_N2_W)[i] = gw;
      WG(W_LN_N2_B)[i] = gb;
    }
  }
  // edge module 2: We2[H,FE] over E rows (gpre in gpe1 buffer)
  {
    const float* gam = WP(W_LN_E2_W);
    const float* bet = WP(W_LN_E2_B);
    for (long u = tid; u < (long)D.H * D.FE; u += NT) {
      int o = (int)(u / D.FE), i = (int)(u % D.FE);
      float acc = 0.f;
      for (int k = 0; k < D.E; ++k)
        acc += P.gpe1[(long)k * D.H + o]
               * (P.xh_e2[(long)k * D.FE + i] * gam[i] + bet[i]);
      WG(W_E2_W)[u] = acc;
    }
    for (int o = tid; o < D.H; o += NT) {
      float acc = 0.f;
      for (int k = 0; k < D.E; ++k) acc += P.gpe1[(long)k * D.H + o];
      WG(W_E2_B)[o] = acc;
    }
    const float* We = WP(W_E2_W);
    for (int i = tid; i < D.FE; i += NT) {
      float gw = 0.f, gb = 0.f;
      for (int k = 0; k < D.E; ++k) {
        float gu = 0.f;
        for (int o = 0; o < D.H; ++o)
          gu += We[o * D.FE + i] * P.gpe1[(long)k * D.H + o];
        gw += gu * P.xh_e2[(long)k * D.FE + i];
        gb += gu;
      }
      WG(W_LN_E2_W)[i] = gw;
      WG(W_LN_E2_B)[i] = gb;
    }
  }
  // reduce MLP 1: Wr1[HID,MSG] over E + N rows (gpre in gpre1_e/gpre1_s)
  {
    const float* gam = WP(W_LN_R1_W);
    const float* bet = WP(W_LN_R1_B);
    for (long u = tid; u < (long)D.HID * D.MSG; u += NT) {
      int o = (int)(u / D.MSG), i = (int)(u % D.MSG);
      float acc = 0.f;
      for (int k = 0; k < D.E; ++k)
        acc += gpre1_e[(long)k * D.HID + o]
               * (P.xh_m1e[(long)k * D.MSG + i] * gam[i] + bet[i]);
      for (int v = 0; v < D.N; ++v)
        acc += gpre1_s[(long)v * D.HID + o]
               * (P.xh_m1s[(long)v * D.MSG + i] * gam[i] + bet[i]);
      WG(W_R1_W)[u] = acc;
    }
    for (int o = tid; o < D.HID; o += NT) {
      float acc = 0.f;
      for (int k = 0; k < D.E; ++k) acc += gpre1_e[(long)k * D.HID + o];
      for (int v = 0; v < D.N; ++v) acc += gpre1_s[(long)v * D.HID + o];
      WG(W_R1_B)[o] = acc;
    }
    const float* Wr = WP(W_R1_W);
    for (int i = tid; i < D.MSG; i += NT) {
      float gw = 0.f, gb = 0.f;
      for (int k = 0; k < D.E; ++k) {
        float gu = 0.f;
        for (int o = 0; o < D.HID; ++o)
          gu += Wr[o * D.MSG + i] * gpre1_e[(long)k * D.HID + o];
        gw += gu * P.xh_m1e[(long)k * D.MSG + i];
        gb += gu;
      }
      for (int v = 0; v < D.N; ++v) {
        float gu = 0.f;
        for (int o = 0; o < D.HID; ++o)
          gu += Wr[o * D.MSG + i] * gpre1_s[(long)v * D.HID + o];
        gw += gu * P.xh_m1s[(long)v * D.MSG + i];
        gb += gu;
      }
      WG(W_LN_R1_W)[i] = gw;
      WG(W_LN_R1_B)[i] = gb;
    }
  }
  // node module 1: Wn1[H,F0] over N rows (gpre in ghn1, relu-masked)
  {
    const float* gam = WP(W_LN_N1_W);
    const float* bet = WP(W_LN_N1_B);
    for (long u = tid; u < (long)D.H * D.F0; u += NT) {
      int o = (int)(u / D.F0), i = (int)(u % D.F0);
      float acc = 0.f;
      for (int v = 0; v < D.N; ++v)
        acc += P.ghn1[(long)v * D.H + o]
               * (P.xh_z1[(long)v * D.F0 + i] * gam[i] + bet[i]);
      WG(W_N1_W)[u] = acc;
    }
    for (int o = tid; o < D.H; o += NT) {
      float acc = 0.f;
      for (int v = 0; v < D.N; ++v) acc += P.ghn1[(long)v * D.H + o];
      WG(W_N1_B)[o] = acc;
    }
    const float* Wn = WP(W_N1_W);
    for (int i = tid; i < D.F0; i += NT) {
      float gw = 0.f, gb = 0.f;
      for (int v = 0; v < D.N; ++v) {
        float gu = 0.f;
        for (int o = 0; o < D.H; ++o)
          gu += Wn[o * D.F0 + i] * P.ghn1[(long)v * D.H + o];
        gw += gu * P.xh_z1[(long)v * D.F0 + i];
        gb += gu;
      }
      WG(W_LN_N1_W)[i] = gw;
      WG(W_LN_N1_B)[i] = gb;
    }
  }
  // edge module 1: We1[H,FE] over E rows (gpre in ghe1)
  {
    const float* gam = WP(W_LN_E1_W);
    const float* bet = WP(W_LN_E1_B);
    for (long u = tid; u < (long)D.H * D.FE; u += NT) {
      int o = (int)(u / D.FE), i = (int)(u % D.FE);
      float acc = 0.f;
      for (int k = 0; k < D.E; ++k)
        acc += P.ghe1[(long)k * D.H + o]
               * (P.xh_e1[(long)k * D.FE + i] * gam[i] + bet[i]);
      WG(W_E1_W)[u] = acc;
    }
    for (int o = tid; o < D.H; o += NT) {
      float acc = 0.f;
      for (int k = 0; k < D.E; ++k) acc += P.ghe1[(long)k * D.H + o];
      WG(W_E1_B)[o] = acc;
    }
    const float* We = WP(W_E1_W);
    for (int i = tid; i < D.FE; i += NT) {
      float gw = 0.f, gb = 0.f;
      for (int k = 0; k < D.E; ++k) {
        float gu = 0.f;
        for (int o = 0; o < D.H; ++o)
          gu += We[o * D.FE + i] * P.ghe1[(long)k * D.H + o];
        gw += gu * P.xh_e1[(long)k * D.FE + i];
        gb += gu;
      }
      WG(W_LN_E1_W)[i] = gw;
      WG(W_LN_E1_B)[i] = gb;
    }
  }
}

// ---------------------------------------------------------------------------
// host bindings
// ---------------------------------------------------------------------------

static void fill_ptrs(CachedPtrs& P, CachedDims& D,
                      std::vector<torch::Tensor>& T,
                      std::vector<double>& fs) {
  TORCH_CHECK((int)T.size() == C_NT, "cached_step: tensor list size");
  D.N = (int)T[C_Z0].size(0);
  D.F0 = (int)T[C_Z0].size(1);
  D.E = (int)T[C_E].size(0);
  D.FE = (int)T[C_E].size(1);
  D.M = (int)T[C_MODEL_NPTR].size(0) - 1;
  D.B = (int)T[C_GF].size(0);
  D.GFin = (int)T[C_GF].size(1);
  D.A = (int)T[C_MASK].size(1);
  D.H = (int)T[C_HN1].size(1);
  D.MSG = 2 * D.H;
  D.HID = (int)T[C_H1].size(1);
  D.OUT = (int)T[C_H2].size(1);
  D.GEMB = (int)T[C_FINAL].size(1) - D.OUT;
  D.FIN = (int)T[C_FINAL].size(1);
  D.FC = (int)T[C_H1P].size(1);
  D.clip = (float)fs[0];
  D.vf_clip = (float)fs[1];
  D.vf_coef = (float)fs[2];
  D.ent_coef = (float)fs[3];
  TORCH_CHECK(D.MSG <= 64 && D.HID <= 64 && D.OUT <= 64 && D.A <= 64
              && D.F0 <= 64 && D.FE <= 64 && D.GFin <= 64 && D.FIN <= 64,
              "cached_step: dims exceed 64");
  P.z0 = T[C_Z0].data_ptr<float>();
  P.e = T[C_E].data_ptr<float>();
  P.src = T[C_SRC].data_ptr<long>();
  P.dst = T[C_DST].data_ptr<long>();
  P.order = T[C_ORDER].data_ptr<long>();
  P.indptr = T[C_INDPTR].data_ptr<long>();
  P.src_order = T[C_SRC_ORDER].data_ptr<long>();
  P.src_indptr = T[C_SRC_INDPTR].data_ptr<long>();
  P.node_model = T[C_NODE_MODEL].data_ptr<long>();
  P.model_nptr = T[C_MODEL_NPTR].data_ptr<long>();
  P.model_ids = T[C_MODEL_IDS].data_ptr<long>();
  P.gf = T[C_GF].data_ptr<float>();
  P.mask = T[C_MASK].data_ptr<float>();
  P.actions = T[C_ACTIONS].data_ptr<long>();
  P.old_logp = T[C_OLD_LOGP].data_ptr<float>();
  P.adv = T[C_ADV].data_ptr<float>();
  P.vtarg = T[C_VTARG].data_ptr<float>();
  P.kl = T[C_KL].data_ptr<float>();
  P.flat_p = T[C_FLAT_P].data_ptr<float>();
  P.flat_g = T[C_FLAT_G].data_ptr<float>();
  P.offs = T[C_OFFS].data_ptr<long>();
  P.hn1 = T[C_HN1].data_ptr<float>();
  P.he1 = T[C_HE1].data_ptr<float>();
  P.re1 = T[C_RE1].data_ptr<float>();
  P.rs1 = T[C_RS1].data_ptr<float>();
  P.h1 = T[C_H1].data_ptr<float>();
  P.hn2 = T[C_HN2].data_ptr<float>();
  P.he2 = T[C_HE2].data_ptr<float>();
  P.re2 = T[C_RE2].data_ptr<float>();
  P.rs2 = T[C_RS2].data_ptr<float>();
  P.h2 = T[C_H2].data_ptr<float>();
  P.pooled = T[C_POOLED].data_ptr<float>();
  P.xh_z1 = T[C_XH_Z1].data_ptr<float>();
  P.xh_e1 = T[C_XH_E1].data_ptr<float>();
  P.xh_m1e = T[C_XH_M1E].data_ptr<float>();
  P.xh_m1s = T[C_XH_M1S].data_ptr<float>();
  P.rst_z1 = T[C_RST_Z1].data_ptr<float>();
  P.rst_e1 = T[C_RST_E1].data_ptr<float>();
  P.rst_m1e = T[C_RST_M1E].data_ptr<float>();
  P.rst_m1s = T[C_RST_M1S].data_ptr<float>();
  P.xh_h2 = T[C_XH_H2].data_ptr<float>();
  P.xh_e2 = T[C_XH_E2].data_ptr<float>();
  P.xh_m2e = T[C_XH_M2E].data_ptr<float>();
  P.xh_m2s = T[C_XH_M2S].data_ptr<float>();
  P.rst_h2 = T[C_RST_H2].data_ptr<float>();
  P.rst_e2 = T[C_RST_E2].data_ptr<float>();
  P.rst_m2e = T[C_RST_M2E].data_ptr<float>();
  P.rst_m2s = T[C_RST_M2S].data_ptr<float>();
  P.xh34 = T[C_XH34].data_ptr<float>();
  P.fin = T[C_FINAL].data_ptr<float>();
  P.h1p = T[C_H1P].data_ptr<float>();
  P.h1v = T[C_H1V].data_ptr<float>();
  P.values = T[C_VALUES].data_ptr<float>();
  P.p = T[C_P].data_ptr<float>();
  P.lp = T[C_LP].data_ptr<float>();
  P.coef = T[C_COEF].data_ptr<float>();
  P.hent = T[C_HENT].data_ptr<float>();
  P.stats = T[C_STATS].data_ptr<float>();
  P.glogits = T[C_GLOGITS].data_ptr<float>();
  P.gvalue = T[C_GVALUE].data_ptr<float>();
  P.gh1p = T[C_GH1P].data_ptr<float>();
  P.gh1v = T[C_GH1V].data_ptr<float>();
  P.gfinal = T[C_GFINAL].data_ptr<float>();
  P.gu34 = T[C_GU34].data_ptr<float>();
  P.gpool = T[C_GPOOL].data_ptr<float>();
  P.gh2 = T[C_GH2].data_ptr<float>();
  P.gme2 = T[C_GME2].data_ptr<float>();
  P.gms2 = T[C_GMS2].data_ptr<float>();
  P.ghn2 = T[C_GHN2].data_ptr<float>();
  P.ghe2 = T[C_GHE2].data_ptr<float>();
  P.gh1b = T[C_GH1B].data_ptr<float>();
  P.gme1 = T[C_GME1].data_ptr<float>();
  P.gms1 = T[C_GMS1].data_ptr<float>();
  P.ghn1 = T[C_GHN1].data_ptr<float>();
  P.ghe1 = T[C_GHE1].data_ptr<float>();
  P.gpn2 = T[C_GPN2].data_ptr<float>();
  P.gpe2 = T[C_GPE2].data_ptr<float>();
  P.gpn1 = T[C_GPN1].data_ptr<float>();
  P.gpe1 = T[C_GPE1].data_ptr<float>();
  P.gpre1_e = T[C_GPRE1E].data_ptr<float>();
  P.gpre1_s = T[C_GPRE1S].data_ptr<float>();
}

void cached_step_fwd(std::vector<torch::Tensor> T, std::vector<double> fs) {
  CachedPtrs P;
  CachedDims D;
  fill_ptrs(P, D, T, fs);
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(cached_step_fwd_kernel, dim3(1), dim3(NTHREADS), 0,
                     stream, P, D);
}

void cached_step_bwd(std::vector<torch::Tensor> T, std::vector<double> fs) {
  CachedPtrs P;
  CachedDims D;
  fill_ptrs(P, D, T, fs);
  hipStream_t stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(cached_step_bwd_kernel, dim3(1), dim3(NTHREADS), 0,
                     stream, P, D);
}
