// cilfw — fused loss / optimizer / metric / herding kernels for gfx950.
// Replaces the reference's ATen softmax+CE (template.py:259), the SoftTarget KD
// composite (utils.py:121-132 -> one fused kernel), torch SGD
// (K10), timm accuracy (K12) and continuum's CPU herding loop (K13).

#include "common.h"

#define NT 256

// ---------------------------------------------- fused CE (+label smoothing) fwd
// logits fp32 (M, C); writes probs (M, C) fp32 and per-row losses (reduced to
// the scalar mean by loss_mean_kernel in a fixed order — deterministic).
// One block per row.

__global__ __launch_bounds__(NT)
void ce_fwd_kernel(const float* __restrict__ logits,
                   const long* __restrict__ targets,
                   float* __restrict__ probs, float* __restrict__ rowloss,
                   int M, int C, float smooth) {
  const int row = blockIdx.x;
  const float* lr = logits + (long)row * C;
  float* pr = probs + (long)row * C;
  __shared__ float red[NT / WAVE];
  // 1) max
  float mx = -3.4e38f;
  for (int c = threadIdx.x; c < C; c += NT) mx = fmaxf(mx, lr[c]);
  for (int o = 32; o > 0; o >>= 1) mx = fmaxf(mx, __shfl_xor(mx, o));
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = mx;
  __syncthreads();
  mx = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
  // 2) sum exp (+ mean logp term for smoothing)
  float se = 0.f;
  for (int c = threadIdx.x; c < C; c += NT) se += __expf(lr[c] - mx);
  for (int o = 32; o > 0; o >>= 1) se += __shfl_xor(se, o);
  __syncthreads();
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = se;
  __syncthreads();
  se = red[0] + red[1] + red[2] + red[3];
  const float lse = __logf(se) + mx;
  // 3) probs + loss
  float sum_logp = 0.f;
  for (int c = threadIdx.x; c < C; c += NT) {
    float logp = lr[c] - lse;
    pr[c] = __expf(logp);
    sum_logp += logp;
  }
  if (smooth > 0.f) {
    for (int o = 32; o > 0; o >>= 1) sum_logp += __shfl_xor(sum_logp, o);
    __syncthreads();
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = sum_logp;
    __syncthreads();
    sum_logp = red[0] + red[1] + red[2] + red[3];
  }
  if (threadIdx.x == 0) {
    float nll = lse - lr[targets[row]];
    rowloss[row] = (1.f - smooth) * nll - smooth * (sum_logp / C);
  }
}

// deterministic mean over per-row losses (single block, fixed tree order)
__global__ __launch_bounds__(NT)
void loss_mean_kernel(const float* __restrict__ rowloss,
                      float* __restrict__ loss, int M, float scale) {
  __shared__ float red[NT];
  float s = 0.f;
  for (int i = threadIdx.x; i < M; i += NT) s += rowloss[i];
  red[threadIdx.x] = s;
  __syncthreads();
  for (int o = NT / 2; o > 0; o >>= 1) {
    if (threadIdx.x < o) red[threadIdx.x] += red[threadIdx.x + o];
    __syncthreads();
  }
  if (threadIdx.x == 0) loss[0] = red[0] * scale;
}

// dlogits = (probs - (1-s)*onehot - s/C) * dloss / M  (fp32)
__global__ __launch_bounds__(NT)
void ce_bwd_kernel(const float* __restrict__ probs,
                   const long* __restrict__ targets,
                   const float* __restrict__ dloss,
                   float* __restrict__ dlogits, int M, int C, float smooth) {
  long i = (long)blockIdx.x * NT + threadIdx.x;
  if (i >= (long)M * C) return;
  int row = (int)(i / C), c = (int)(i % C);
  float g = probs[i] - smooth / C;
  if (c == (int)targets[row]) g -= (1.f - smooth);
  dlogits[i] = g * dloss[0] / M;
}

// -------------------------------------------------- fused KD (SoftTarget) fwd
// s,t fp32 logits (M, C); ps/pt fp32 softmax(x/T); loss += T^2/M * KL(pt||ps).

__global__ __launch_bounds__(NT)
void kd_fwd_kernel(const float* __restrict__ s, const float* __restrict__ tt,
                   float* __restrict__ ps, float* __restrict__ pt,
                   float* __restrict__ loss, int M, int C, float T) {
  const int row = blockIdx.x;
  __shared__ float red[NT / WAVE];
  const float* rows[2] = {s + (long)row * C, tt + (long)row * C};
  float* outs[2] = {ps + (long)row * C, pt + (long)row * C};
  float lse[2];
#pragma unroll
  for (int which = 0; which < 2; ++which) {
    const float* lr = rows[which];
    float mx = -3.4e38f;
    for (int c = threadIdx.x; c < C; c += NT) mx = fmaxf(mx, lr[c] / T);
    for (int o = 32; o > 0; o >>= 1) mx = fmaxf(mx, __shfl_xor(mx, o));
    __syncthreads();
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = mx;
    __syncthreads();
    mx = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
    float se = 0.f;
    for (int c = threadIdx.x; c < C; c += NT) se += __expf(lr[c] / T - mx);
    for (int o = 32; o > 0; o >>= 1) se += __shfl_xor(se, o);
    __syncthreads();
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = se;
    __syncthreads();
    se = red[0] + red[1] + red[2] + red[3];
    lse[which] = __logf(se) + mx;
    for (int c = threadIdx.x; c < C; c += NT)
      outs[which][c] = __expf(lr[c] / T - lse[which]);
  }
  // loss_row = sum pt * (logpt - logps)
  float l = 0.f;
  const float* sr = rows[0];
  const float* tr = rows[1];
  for (int c = threadIdx.x; c < C; c += NT) {
    float logps = sr[c] / T - lse[0];
    float logpt = tr[c] / T - lse[1];
    l += __expf(logpt) * (logpt - logps);
  }
  for (int o = 32; o > 0; o >>= 1) l += __shfl_xor(l, o);
  __syncthreads();
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = l;
  __syncthreads();
  if (threadIdx.x == 0) {
    l = red[0] + red[1] + red[2] + red[3];
    loss[row] = l * T * T;  // per-row; loss_mean_kernel reduces
  }
}

// ds = (ps - pt) * T / M * dloss
__global__ __launch_bounds__(NT)
void kd_bwd_kernel(const float* __restrict__ ps, const float* __restrict__ pt,
                   const float* __restrict__ dloss, float* __restrict__ ds,
                   long total, int M, float T) {
  long i = (long)blockIdx.x * NT + threadIdx.x;
  if (i >= total) return;
  ds[i] = (ps[i] - pt[i]) * (T / M) * dloss[0];
}

// ------------------------------------------- fully fused WA loss (CE + KD)
// One kernel pair for the whole WA training objective: CE(+label smoothing)
// over all C classes + lambda * SoftTarget KD over the first Ck (teacher)
// classes, reading the bf16 logits DIRECTLY (no fp32 cast round-trips) and
// emitting bf16 dlogits for the classifier backward. One block per row.

__global__ __launch_bounds__(NT)
void wa_loss_fwd_kernel(const bf16_t* __restrict__ slog,
                        const bf16_t* __restrict__ tlog,
                        const long* __restrict__ targets,
                        float* __restrict__ probs,   // (M, C)
                        float* __restrict__ ps,      // (M, Ck)
                        float* __restrict__ pt,      // (M, Ck)
                        float* __restrict__ rl_ce, float* __restrict__ rl_kd,
                        int M, int C, int Ck, float smooth, float T) {
  const int row = blockIdx.x;
  const bf16_t* sr = slog + (long)row * C;
  __shared__ float red[NT / WAVE];

  // ---- CE over all C ----
  float mx = -3.4e38f;
  for (int c = threadIdx.x; c < C; c += NT) mx = fmaxf(mx, bf2f(sr[c]));
  for (int o = 32; o > 0; o >>= 1) mx = fmaxf(mx, __shfl_xor(mx, o));
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = mx;
  __syncthreads();
  mx = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
  float se = 0.f;
  for (int c = threadIdx.x; c < C; c += NT) se += __expf(bf2f(sr[c]) - mx);
  for (int o = 32; o > 0; o >>= 1) se += __shfl_xor(se, o);
  __syncthreads();
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = se;
  __syncthreads();
  se = red[0] + red[1] + red[2] + red[3];
  const float lse = __logf(se) + mx;
  float sum_logp = 0.f;
  for (int c = threadIdx.x; c < C; c += NT) {
    float logp = bf2f(sr[c]) - lse;
    probs[(long)row * C + c] = __expf(logp);
    sum_logp += logp;
  }
  if (smooth > 0.f) {
    for (int o = 32; o > 0; o >>= 1) sum_logp += __shfl_xor(sum_logp, o);
    __syncthreads();
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = sum_logp;
    __syncthreads();
    sum_logp = red[0] + red[1] + red[2] + red[3];
  }
  if (threadIdx.x == 0) {
    float nll = lse - bf2f(sr[targets[row]]);
    rl_ce[row] = (1.f - smooth) * nll - smooth * (sum_logp / C);
  }

  if (Ck <= 0) {
    if (threadIdx.x == 0) rl_kd[row] = 0.f;
    return;
  }

  // ---- KD over the first Ck columns ----
  const bf16_t* tr = tlog + (long)row * Ck;
  float lsek[2];
#pragma unroll
  for (int which = 0; which < 2; ++which) {
    const bf16_t* lr = which == 0 ? sr : tr;
    float m2 = -3.4e38f;
    for (int c = threadIdx.x; c < Ck; c += NT)
      m2 = fmaxf(m2, bf2f(lr[c]) / T);
    for (int o = 32; o > 0; o >>= 1) m2 = fmaxf(m2, __shfl_xor(m2, o));
    __syncthreads();
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = m2;
    __syncthreads();
    m2 = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
    float s2 = 0.f;
    for (int c = threadIdx.x; c < Ck; c += NT)
      s2 += __expf(bf2f(lr[c]) / T - m2);
    for (int o = 32; o > 0; o >>= 1) s2 += __shfl_xor(s2, o);
    __syncthreads();
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = s2;
    __syncthreads();
    s2 = red[0] + red[1] + red[2] + red[3];
    lsek[which] = __logf(s2) + m2;
    float* out = which == 0 ? ps : pt;
    for (int c = threadIdx.x; c < Ck; c += NT)
      out[(long)row * Ck + c] = __expf(bf2f(lr[c]) / T - lsek[which]);
  }
  float l = 0.f;
  for (int c = threadIdx.x; c < Ck; c += NT) {
    float logps = bf2f(sr[c]) / T - lsek[0];
    float logpt = bf2f(tr[c]) / T - lsek[1];
    l += __expf(logpt) * (logpt - logps);
  }
  for (int o = 32; o > 0; o >>= 1) l += __shfl_xor(l, o);
  __syncthreads();
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = l;
  __syncthreads();
  if (threadIdx.x == 0) {
    l = red[0] + red[1] + red[2] + red[3];
    rl_kd[row] = l * T * T;
  }
}

// out[0]=mean(ce), out[1]=mean(kd), out[2]=mean(ce)+lam*mean(kd)
__global__ __launch_bounds__(NT)
void loss_mean3_kernel(const float* __restrict__ rl_ce,
                       const float* __restrict__ rl_kd,
                       float* __restrict__ out, int M, float lam) {
  __shared__ float red[2][NT];
  float a = 0.f, b = 0.f;
  for (int i = threadIdx.x; i < M; i += NT) {
    a += rl_ce[i];
    b += rl_kd[i];
  }
  red[0][threadIdx.x] = a;
  red[1][threadIdx.x] = b;
  __syncthreads();
  for (int o = NT / 2; o > 0; o >>= 1) {
    if (threadIdx.x < o) {
      red[0][threadIdx.x] += red[0][threadIdx.x + o];
      red[1][threadIdx.x] += red[1][threadIdx.x + o];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    out[0] = red[0][0] / M;
    out[1] = red[1][0] / M;
    out[2] = out[0] + lam * out[1];
  }
}

// dlogits(bf16) = dtotal * [ dCE + lam*dKD ]   (dtotal read from device)
__global__ __launch_bounds__(NT)
void wa_loss_bwd_kernel(const float* __restrict__ probs,
                        const float* __restrict__ ps,
                        const float* __restrict__ pt,
                        const long* __restrict__ targets,
                        const float* __restrict__ dtotal,
                        bf16_t* __restrict__ dlogits, int M, int C, int Ck,
                        float smooth, float T, float lam) {
  long i = (long)blockIdx.x * NT + threadIdx.x;
  if (i >= (long)M * C) return;
  int row = (int)(i / C), c = (int)(i % C);
  float g = probs[i] - smooth / C;
  if (c == (int)targets[row]) g -= (1.f - smooth);
  if (c < Ck) {
    long j = (long)row * Ck + c;
    g += lam * T * (ps[j] - pt[j]);
  }
  dlogits[i] = f2bf(g * dtotal[0] / M);
}

// ------------------------------------------------------------------- fused SGD
// g = grad + wd*p; m = mu*m + g; p -= lr*m   — one kernel over the flat buffers.

// lr arrives through DEVICE memory (lr_dev) so a hipGraph-captured step can
// be re-used across epochs: the cosine scheduler writes the scalar once per
// epoch instead of forcing a re-capture (the lr would otherwise be baked in)
__global__ __launch_bounds__(NT)
void sgd_kernel(float* __restrict__ p, const float* __restrict__ g,
                float* __restrict__ m, bf16_t* __restrict__ pb, long n,
                const float* __restrict__ lr_dev, float mu, float wd) {
  const float lr = lr_dev[0];
  long i0 = ((long)blockIdx.x * NT + threadIdx.x) * 4;
  if (i0 >= n) return;
  if (i0 + 4 <= n) {
    float4 pv = *(float4*)&p[i0];
    float4 gv = *(const float4*)&g[i0];
    float4 mv = *(float4*)&m[i0];
    float* pe = (float*)&pv;
    const float* ge = (const float*)&gv;
    float* me = (float*)&mv;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gg = ge[j] + wd * pe[j];
      me[j] = mu * me[j] + gg;
      pe[j] -= lr * me[j];
    }
    *(float4*)&p[i0] = pv;
    *(float4*)&m[i0] = mv;
    if (pb != nullptr) {  // keep the bf16 compute mirror in sync in-kernel
      bf16_t o[4] = {f2bf(pe[0]), f2bf(pe[1]), f2bf(pe[2]), f2bf(pe[3])};
      *(int2*)&pb[i0] = *(int2*)o;
    }
  } else {
    for (long i = i0; i < n; ++i) {
      float gg = g[i] + wd * p[i];
      m[i] = mu * m[i] + gg;
      p[i] -= lr * m[i];
      if (pb != nullptr) pb[i] = f2bf(p[i]);
    }
  }
}

// -------------------------------------------------------------- top-k correct
// counts[k-1] += 1 if target is within the top-k logits (k = 1..maxk).
// One block per row; maxk passes of masked argmax (maxk <= 8).

__global__ __launch_bounds__(NT)
void topk_kernel(const float* __restrict__ logits,
                 const long* __restrict__ targets,
                 long* __restrict__ counts, int M, int C, int maxk) {
  const int row = blockIdx.x;
  const float* lr = logits + (long)row * C;
  const int tgt = (int)targets[row];
  __shared__ float rv[NT / WAVE];
  __shared__ int ri[NT / WAVE];
  __shared__ int found_k;
  if (threadIdx.x == 0) found_k = -1;
  __shared__ int excluded[8];
  __syncthreads();
  for (int k = 0; k < maxk; ++k) {
    float best = -3.4e38f;
    int bi = INT32_MAX;
    for (int c = threadIdx.x; c < C; c += NT) {
      bool skip = false;
      for (int e = 0; e < k; ++e) skip |= (excluded[e] == c);
      if (skip) continue;
      float v = lr[c];
      // torch.topk tie-break: lower index wins
      if (v > best || (v == best && c < bi)) { best = v; bi = c; }
    }
    for (int o = 32; o > 0; o >>= 1) {
      float ov = __shfl_xor(best, o);
      int oi = __shfl_xor(bi, o);
      if (ov > best || (ov == best && oi < bi)) { best = ov; bi = oi; }
    }
    __syncthreads();
    if ((threadIdx.x & 63) == 0) {
      rv[threadIdx.x >> 6] = best;
      ri[threadIdx.x >> 6] = bi;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      for (int w = 1; w < NT / WAVE; ++w)
        if (rv[w] > best || (rv[w] == best && ri[w] < bi)) {
          best = rv[w];
          bi = ri[w];
        }
      excluded[k] = bi;
      if (bi == tgt && found_k < 0) found_k = k;
    }
    __syncthreads();
  }
  if (threadIdx.x == 0 && found_k >= 0)
    for (int k = found_k; k < maxk; ++k)
      atomicAdd((unsigned long long*)&counts[k], 1ull);
}

// ------------------------------------------------------------ herding select
// Greedy barycenter herding, single persistent block (features (n, D) fp32).
// Each iteration k: argmin_i || (sum_sel + f_i)/(k+1) - mu ||^2 over unselected,
// ties to the lowest index (matches torch.argmin in the CPU oracle).

__global__ __launch_bounds__(NT)
void herding_kernel(const float* __restrict__ f, const float* __restrict__ mu,
                    long* __restrict__ order, int n, int D, int m) {
  extern __shared__ float sh[];         // [D] sum_sel + [n] selected flags
  float* sum_sel = sh;
  float* selmark = sh + D;
  for (int d = threadIdx.x; d < D; d += NT) sum_sel[d] = 0.f;
  for (int i = threadIdx.x; i < n; i += NT) selmark[i] = 0.f;
  __shared__ float rv[NT / WAVE];
  __shared__ int ri[NT / WAVE];
  __shared__ int chosen;
  __syncthreads();
  for (int k = 0; k < m; ++k) {
    const float inv = 1.f / (k + 1);
    float best = 3.4e38f;
    int bi = INT32_MAX;
    for (int i = threadIdx.x; i < n; i += NT) {
      if (selmark[i] != 0.f) continue;
      const float* fi = f + (long)i * D;
      float d2 = 0.f;
      for (int d = 0; d < D; ++d) {
        float t = (sum_sel[d] + fi[d]) * inv - mu[d];
        d2 += t * t;
      }
      if (d2 < best || (d2 == best && i < bi)) { best = d2; bi = i; }
    }
    for (int o = 32; o > 0; o >>= 1) {
      float ov = __shfl_xor(best, o);
      int oi = __shfl_xor(bi, o);
      if (ov < best || (ov == best && oi < bi)) { best = ov; bi = oi; }
    }
    __syncthreads();
    if ((threadIdx.x & 63) == 0) {
      rv[threadIdx.x >> 6] = best;
      ri[threadIdx.x >> 6] = bi;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      for (int w = 1; w < NT / WAVE; ++w)
        if (rv[w] < best || (rv[w] == best && ri[w] < bi)) {
          best = rv[w];
          bi = ri[w];
        }
      chosen = bi;
      order[k] = bi;
      selmark[bi] = 1.f;
    }
    __syncthreads();
    const float* fc = f + (long)chosen * D;
    for (int d = threadIdx.x; d < D; d += NT) sum_sel[d] += fc[d];
    __syncthreads();
  }
}

// batched herding: one BLOCK per class, all classes' greedy selections run
// CONCURRENTLY (the per-class loop is inherently sequential in m, so the
// parallelism axis is classes — SURVEY §7 hard-parts item). Layout: features
// flat (sum n_c, D) grouped by class with offsets foff[c]; per-class means
// mu[c]; outputs ranked indices (LOCAL to the class) at ooff[c].
__global__ __launch_bounds__(NT)
void herding_batch_kernel(const float* __restrict__ fall,
                          const int* __restrict__ foff,
                          const float* __restrict__ muall,
                          const int* __restrict__ ooff,
                          const int* __restrict__ mvec,
                          long* __restrict__ order, int D) {
  const int cls = blockIdx.x;
  const float* f = fall + (long)foff[cls] * D;
  const float* mu = muall + (long)cls * D;
  long* out = order + ooff[cls];
  const int n = foff[cls + 1] - foff[cls];
  const int m = mvec[cls];
  extern __shared__ float sh[];
  float* sum_sel = sh;
  float* selmark = sh + D;
  for (int d = threadIdx.x; d < D; d += NT) sum_sel[d] = 0.f;
  for (int i = threadIdx.x; i < n; i += NT) selmark[i] = 0.f;
  __shared__ float rv[NT / WAVE];
  __shared__ int ri[NT / WAVE];
  __shared__ int chosen;
  __syncthreads();
  for (int k = 0; k < m; ++k) {
    const float inv = 1.f / (k + 1);
    float best = 3.4e38f;
    int bi = INT32_MAX;
    for (int i = threadIdx.x; i < n; i += NT) {
      if (selmark[i] != 0.f) continue;
      const float* fi = f + (long)i * D;
      float d2 = 0.f;
      for (int d = 0; d < D; ++d) {
        float t = (sum_sel[d] + fi[d]) * inv - mu[d];
        d2 += t * t;
      }
      if (d2 < best || (d2 == best && i < bi)) { best = d2; bi = i; }
    }
    for (int o = 32; o > 0; o >>= 1) {
      float ov = __shfl_xor(best, o);
      int oi = __shfl_xor(bi, o);
      if (ov < best || (ov == best && oi < bi)) { best = ov; bi = oi; }
    }
    __syncthreads();
    if ((threadIdx.x & 63) == 0) {
      rv[threadIdx.x >> 6] = best;
      ri[threadIdx.x >> 6] = bi;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      for (int w = 1; w < NT / WAVE; ++w)
        if (rv[w] < best || (rv[w] == best && ri[w] < bi)) {
          best = rv[w];
          bi = ri[w];
        }
      chosen = bi;
      out[k] = bi;
      selmark[bi] = 1.f;
    }
    __syncthreads();
    const float* fc = f + (long)chosen * D;
    for (int d = threadIdx.x; d < D; d += NT) sum_sel[d] += fc[d];
    __syncthreads();
  }
}

// ============================== launchers ==============================

extern "C" {

void cilfw_ce_fwd(const void* logits, const void* targets, void* probs,
                  void* loss, void* rowloss, int M, int C, float smooth,
                  void* stream) {
  hipLaunchKernelGGL(ce_fwd_kernel, dim3(M), dim3(NT), 0,
                     (hipStream_t)stream, (const float*)logits,
                     (const long*)targets, (float*)probs, (float*)rowloss, M,
                     C, smooth);
  hipLaunchKernelGGL(loss_mean_kernel, dim3(1), dim3(NT), 0,
                     (hipStream_t)stream, (const float*)rowloss,
                     (float*)loss, M, 1.f / M);
}

void cilfw_ce_bwd(const void* probs, const void* targets, const void* dloss,
                  void* dlogits, int M, int C, float smooth, void* stream) {
  long total = (long)M * C;
  hipLaunchKernelGGL(ce_bwd_kernel, dim3((int)cdiv((long)total, (long)NT)),
                     dim3(NT), 0, (hipStream_t)stream, (const float*)probs,
                     (const long*)targets, (const float*)dloss,
                     (float*)dlogits, M, C, smooth);
}

void cilfw_kd_fwd(const void* s, const void* t, void* ps, void* pt,
                  void* loss, void* rowloss, int M, int C, float T,
                  void* stream) {
  hipLaunchKernelGGL(kd_fwd_kernel, dim3(M), dim3(NT), 0,
                     (hipStream_t)stream, (const float*)s, (const float*)t,
                     (float*)ps, (float*)pt, (float*)rowloss, M, C, T);
  hipLaunchKernelGGL(loss_mean_kernel, dim3(1), dim3(NT), 0,
                     (hipStream_t)stream, (const float*)rowloss,
                     (float*)loss, M, 1.f / M);
}

void cilfw_kd_bwd(const void* ps, const void* pt, const void* dloss, void* ds,
                  int M, int C, float T, void* stream) {
  long total = (long)M * C;
  hipLaunchKernelGGL(kd_bwd_kernel, dim3((int)cdiv((long)total, (long)NT)),
                     dim3(NT), 0, (hipStream_t)stream, (const float*)ps,
                     (const float*)pt, (const float*)dloss, (float*)ds, total,
                     M, T);
}

void cilfw_sgd_step(void* p, const void* g, void* m, void* pb, long n,
                    const void* lr_dev, float mu, float wd, void* stream) {
  long blocks = cdiv((long)n, (long)NT * 4);
  hipLaunchKernelGGL(sgd_kernel, dim3((int)blocks), dim3(NT), 0,
                     (hipStream_t)stream, (float*)p, (const float*)g,
                     (float*)m, (bf16_t*)pb, n, (const float*)lr_dev, mu, wd);
}

void cilfw_topk_correct(const void* logits, const void* targets, void* counts,
                        int M, int C, int maxk, void* stream) {
  (void)hipMemsetAsync(counts, 0, maxk * sizeof(long), (hipStream_t)stream);
  hipLaunchKernelGGL(topk_kernel, dim3(M), dim3(NT), 0, (hipStream_t)stream,
                     (const float*)logits, (const long*)targets,
                     (long*)counts, M, C, maxk);
}

void cilfw_herding_select_batch(const void* fall, const void* foff,
                                const void* mu, const void* ooff,
                                const void* mvec, void* order, int nclasses,
                                int D, int max_n, void* stream) {
  size_t shmem = (size_t)(D + max_n) * sizeof(float);
  hipLaunchKernelGGL(herding_batch_kernel, dim3(nclasses), dim3(NT), shmem,
                     (hipStream_t)stream, (const float*)fall,
                     (const int*)foff, (const float*)mu, (const int*)ooff,
                     (const int*)mvec, (long*)order, D);
}

void cilfw_herding_select(const void* f, const void* mu, void* order, int n,
                          int D, int m, void* stream) {
  size_t shmem = (D + n) * sizeof(float);
  hipLaunchKernelGGL(herding_kernel, dim3(1), dim3(NT), shmem,
                     (hipStream_t)stream, (const float*)f, (const float*)mu,
                     (long*)order, n, D, m);
}

void cilfw_wa_loss_fwd(const void* slog, const void* tlog,
                       const void* targets, void* probs, void* ps, void* pt,
                       void* rowloss2, void* out3, int M, int C, int Ck,
                       float smooth, float T, float lam, void* stream) {
  float* rl_ce = (float*)rowloss2;
  float* rl_kd = rl_ce + M;
  hipLaunchKernelGGL(wa_loss_fwd_kernel, dim3(M), dim3(NT), 0,
                     (hipStream_t)stream, (const bf16_t*)slog,
                     (const bf16_t*)tlog, (const long*)targets, (float*)probs,
                     (float*)ps, (float*)pt, rl_ce, rl_kd, M, C, Ck, smooth,
                     T);
  hipLaunchKernelGGL(loss_mean3_kernel, dim3(1), dim3(NT), 0,
                     (hipStream_t)stream, rl_ce, rl_kd, (float*)out3, M, lam);
}

void cilfw_wa_loss_bwd(const void* probs, const void* ps, const void* pt,
                       const void* targets, const void* dtotal, void* dlogits,
                       int M, int C, int Ck, float smooth, float T, float lam,
                       void* stream) {
  long total = (long)M * C;
  hipLaunchKernelGGL(wa_loss_bwd_kernel,
                     dim3((int)cdiv((long)total, (long)NT)), dim3(NT), 0,
                     (hipStream_t)stream, (const float*)probs,
                     (const float*)ps, (const float*)pt,
                     (const long*)targets, (const float*)dtotal,
                     (bf16_t*)dlogits, M, C, Ck, smooth, T, lam);
}

}  // extern "C"
