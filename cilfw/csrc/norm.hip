// cilfw — BatchNorm (fused ReLU), add+ReLU, DownsampleA, global-avg-pool and
// max-pool kernels for gfx950. NHWC bf16 activations, fp32 statistics/params.
// Replaces the reference's cuDNN BN + ATen elementwise kernels
// (SURVEY.md §2.3 K3/K4/K5/K6).

#include "common.h"

#define NT 256

// ---------------------------------------------------------------- BN statistics
// pass 1: per-channel sum & sumsq via per-block partials + fp32 atomics.
// x viewed as (M, C); grid.x covers C/64, grid.y covers row-chunks.

// 256 threads = 32 row-lanes x 8 channel-groups of 8 (b128 loads); the block
// covers a 64-channel slice (C % 8 == 0 in all cilfw models; C < 64 handled by
// per-group bounds) over rows_per_blk rows, partials reduced in LDS, then one
// fp32 atomic per (channel, quantity) per block.
__global__ __launch_bounds__(NT)
void bn_sums_kernel(const bf16_t* __restrict__ x, float* __restrict__ part,
                    long M, int C, int rows_per_blk) {
  // part: [gridDim.y][2][C] fp32 partials (deterministic tree reduce follows;
  // the reference runs under cudnn.deterministic — cilfw matches: no fp32
  // atomics anywhere in the training step)
  const int cg = threadIdx.x & 7;          // channel group (8 ch)
  const int rl = threadIdx.x >> 3;         // row lane 0..31
  const int c8 = blockIdx.x * 64 + cg * 8;
  const bool active = c8 + 8 <= C || c8 < C;
  long r0 = (long)blockIdx.y * rows_per_blk + rl;
  long r1 = min((long)(blockIdx.y + 1) * rows_per_blk, M);
  float s[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float q[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  if (c8 + 8 <= C) {
    for (long r = r0; r < r1; r += 32) {
      int4 v = *(const int4*)&x[r * C + c8];
      const bf16_t* e = (const bf16_t*)&v;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(e[j]);
        s[j] += f;
        q[j] += f * f;
      }
    }
  } else if (active) {  // ragged channel tail
    for (long r = r0; r < r1; r += 32)
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (c8 + j < C) {
          float f = bf2f(x[r * C + c8 + j]);
          s[j] += f;
          q[j] += f * f;
        }
  }
  __shared__ float red[2][32][64];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[0][rl][cg * 8 + j] = s[j];
    red[1][rl][cg * 8 + j] = q[j];
  }
  __syncthreads();
  // 128 threads each own one (quantity, channel): serial sum over 32 row lanes
  const int qi = threadIdx.x >> 7;         // 0..1 (threads 0..255 -> 2x128)
  const int cc = threadIdx.x & 127;
  if (cc < 64 && blockIdx.x * 64 + cc < C) {
    float acc = 0.f;
#pragma unroll 8
    for (int r = 0; r < 32; ++r) acc += red[qi][r][cc];
    part[((long)blockIdx.y * 2 + qi) * C + blockIdx.x * 64 + cc] = acc;
  }
}

// deterministic slab reduce: out[i] = sum_s ws[s][i]. One WAVE per output
// element — 64 lanes stride the slab axis in parallel, then a fixed-order
// shfl tree (ns can be ~512: a thread-serial loop would be pure latency).
// out2/split: elements i >= split land in out2[i-split] — lets the backward
// write dgamma and dbeta straight into two separate flat-grad slots
// (grad-sink delivery; cilfw/distributed/ddp.py)
__global__ __launch_bounds__(NT)
void bn_reduce_slabs_kernel(const float* __restrict__ ws,
                            float* __restrict__ out, int ns, long len,
                            float* __restrict__ out2, long split) {
  const int lane = threadIdx.x & 63;
  long i = (long)blockIdx.x * (NT / WAVE) + (threadIdx.x >> 6);
  if (i >= len) return;
  float a = 0.f;
  for (int s = lane; s < ns; s += WAVE) a += ws[(long)s * len + i];
  for (int o = 32; o > 0; o >>= 1) a += __shfl_xor(a, o);
  if (lane == 0) {
    if (out2 != nullptr && i >= split) out2[i - split] = a;
    else out[i] = a;
  }
}

// fused slab-reduce + finalize for TRAINING: one wave per channel reduces the
// [gy][2][C] partials and computes mean/invstd/running-stat update directly
// (drops one launch per training BN forward).
__global__ __launch_bounds__(NT)
void bn_reduce_finalize_kernel(const float* __restrict__ part,
                               float* __restrict__ mean,
                               float* __restrict__ invstd,
                               float* __restrict__ running_mean,
                               float* __restrict__ running_var,
                               int gy, long M, int C, float momentum,
                               float eps) {
  const int lane = threadIdx.x & 63;
  const int c = blockIdx.x * (NT / WAVE) + (threadIdx.x >> 6);
  if (c >= C) return;
  float s = 0.f, q = 0.f;
  for (int g = lane; g < gy; g += WAVE) {
    s += part[((long)g * 2) * C + c];
    q += part[((long)g * 2 + 1) * C + c];
  }
  for (int o = 32; o > 0; o >>= 1) {
    s += __shfl_xor(s, o);
    q += __shfl_xor(q, o);
  }
  if (lane == 0) {
    float mu = s / (float)M;
    float var = fmaxf(q / (float)M - mu * mu, 0.f);
    mean[c] = mu;
    invstd[c] = rsqrtf(var + eps);
    float unbiased = var * (float)M / (float)max(M - 1, 1L);
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mu;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// pass 2: finalize mean/invstd (+ running stats update, training only)
__global__ void bn_finalize_kernel(const float* __restrict__ sum,
                                   const float* __restrict__ sumsq,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   long M, int C, float momentum, float eps,
                                   int training) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  if (training) {
    float mu = sum[c] / (float)M;
    float var = fmaxf(sumsq[c] / (float)M - mu * mu, 0.f);
    mean[c] = mu;
    invstd[c] = rsqrtf(var + eps);
    float unbiased = var * (float)M / (float)max(M - 1, 1L);
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mu;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  } else {
    mean[c] = running_mean[c];
    invstd[c] = rsqrtf(running_var[c] + eps);
  }
}

// pass 3: y = x*sc + sh (+res, +relu), sc = gamma*invstd,
// sh = beta - mean*gamma*invstd. Folding BN to scale/shift lets the per-
// thread channel params come from LDS as FOUR ds_read_b128 instead of 32
// scalar reads (the scalar form measured LDS-bound at ~2.4 TB/s on the
// ResNet-50 stem shapes — less than half the HBM roofline).
__global__ __launch_bounds__(NT)
void bn_apply_kernel(const bf16_t* __restrict__ x, bf16_t* __restrict__ y,
                     const bf16_t* __restrict__ res,
                     const float* __restrict__ gamma,
                     const float* __restrict__ beta,
                     const float* __restrict__ mean,
                     const float* __restrict__ invstd,
                     long total, int C, int relu) {
  extern __shared__ float params[];  // [2][C]: scale, shift
  for (int c = threadIdx.x; c < C; c += NT) {
    float sc = gamma[c] * invstd[c];
    params[c] = sc;
    params[C + c] = beta[c] - mean[c] * sc;
  }
  __syncthreads();
  long i0 = ((long)blockIdx.x * NT + threadIdx.x) * 8;
  if (i0 + 8 > total) {
    for (long i = i0; i < total; ++i) {
      int c = (int)(i % C);
      float v = bf2f(x[i]) * params[c] + params[C + c];
      if (res != nullptr) v += bf2f(res[i]);
      if (relu) v = fmaxf(v, 0.f);
      y[i] = f2bf(v);
    }
    return;
  }
  int4 xv = *(const int4*)&x[i0];
  bf16_t* xe = (bf16_t*)&xv;
  int4 rv;
  const bf16_t* re = (const bf16_t*)&rv;
  if (res != nullptr) rv = *(const int4*)&res[i0];
  bf16_t out[8];
  int c0 = (int)(i0 % C);  // C % 8 == 0 for all cilfw models (8-aligned)
  float4 sc0 = *(const float4*)&params[c0];
  float4 sc1 = *(const float4*)&params[c0 + 4];
  float4 sh0 = *(const float4*)&params[C + c0];
  float4 sh1 = *(const float4*)&params[C + c0 + 4];
  float scv[8] = {sc0.x, sc0.y, sc0.z, sc0.w, sc1.x, sc1.y, sc1.z, sc1.w};
  float shv[8] = {sh0.x, sh0.y, sh0.z, sh0.w, sh1.x, sh1.y, sh1.z, sh1.w};
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float v = bf2f(xe[j]) * scv[j] + shv[j];
    if (res != nullptr) v += bf2f(re[j]);
    if (relu) v = fmaxf(v, 0.f);
    out[j] = f2bf(v);
  }
  *(int4*)&y[i0] = *(int4*)out;
}

// backward pass 1: dbeta = sum dy', dgamma = sum dy'*xhat  (dy' relu-masked)
__global__ __launch_bounds__(NT)
void bn_bwd_sums_kernel(const bf16_t* __restrict__ dy,
                        const bf16_t* __restrict__ x,
                        const bf16_t* __restrict__ y,
                        const float* __restrict__ mean,
                        const float* __restrict__ invstd,
                        float* __restrict__ part,
                        long M, int C, int rows_per_blk, int relu) {
  const int cg = threadIdx.x & 7;
  const int rl = threadIdx.x >> 3;
  const int c8 = blockIdx.x * 64 + cg * 8;
  float mu[8], is[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int c = c8 + j;
    mu[j] = (c < C) ? mean[c] : 0.f;
    is[j] = (c < C) ? invstd[c] : 0.f;
  }
  long r0 = (long)blockIdx.y * rows_per_blk + rl;
  long r1 = min((long)(blockIdx.y + 1) * rows_per_blk, M);
  float sg[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float sb[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  if (c8 + 8 <= C) {
    for (long r = r0; r < r1; r += 32) {
      long i = r * C + c8;
      int4 dv = *(const int4*)&dy[i];
      int4 xv = *(const int4*)&x[i];
      const bf16_t* de = (const bf16_t*)&dv;
      const bf16_t* xe = (const bf16_t*)&xv;
      if (relu) {
        int4 yv = *(const int4*)&y[i];
        const bf16_t* ye = (const bf16_t*)&yv;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float g = (bf2f(ye[j]) > 0.f) ? bf2f(de[j]) : 0.f;
          sb[j] += g;
          sg[j] += g * (bf2f(xe[j]) - mu[j]) * is[j];
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float g = bf2f(de[j]);
          sb[j] += g;
          sg[j] += g * (bf2f(xe[j]) - mu[j]) * is[j];
        }
      }
    }
  } else if (c8 < C) {
    for (long r = r0; r < r1; r += 32)
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (c8 + j < C) {
          long i = r * C + c8 + j;
          float g = bf2f(dy[i]);
          if (relu && bf2f(y[i]) <= 0.f) g = 0.f;
          sb[j] += g;
          sg[j] += g * (bf2f(x[i]) - mu[j]) * is[j];
        }
  }
  __shared__ float red[2][32][64];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[0][rl][cg * 8 + j] = sg[j];
    red[1][rl][cg * 8 + j] = sb[j];
  }
  __syncthreads();
  const int qi = threadIdx.x >> 7;
  const int cc = threadIdx.x & 127;
  if (cc < 64 && blockIdx.x * 64 + cc < C) {
    float acc = 0.f;
#pragma unroll 8
    for (int r = 0; r < 32; ++r) acc += red[qi][r][cc];
    part[((long)blockIdx.y * 2 + qi) * C + blockIdx.x * 64 + cc] = acc;
  }
}

// backward pass 2: dx = (gamma*invstd/M) * (M*dy' - dbeta - xhat*dgamma)
__global__ __launch_bounds__(NT)
void bn_bwd_apply_kernel(const bf16_t* __restrict__ dy,
                         const bf16_t* __restrict__ x,
                         const bf16_t* __restrict__ y,
                         bf16_t* __restrict__ dx,
                         bf16_t* __restrict__ dres,
                         const float* __restrict__ gamma,
                         const float* __restrict__ mean,
                         const float* __restrict__ invstd,
                         const float* __restrict__ dgamma,
                         const float* __restrict__ dbeta,
                         long total, long M, int C, int relu, int training) {
  // folded: dx = P*dy' + Q*x + R  (P = gamma*invstd; training adds the
  // batch-stat correction terms; eval has Q = R = 0) — params come from LDS
  // as float4 pairs like bn_apply (the 5-array scalar form was LDS-bound)
  extern __shared__ float params[];  // [3][C]: P, Q, R
  const float rM = 1.f / (float)M;
  for (int c = threadIdx.x; c < C; c += NT) {
    float P = gamma[c] * invstd[c];
    float Q = 0.f, R = 0.f;
    if (training) {
      float A = P * rM;
      Q = -A * dgamma[c] * invstd[c];
      R = A * (dgamma[c] * invstd[c] * mean[c] - dbeta[c]);
    }
    params[c] = P;
    params[C + c] = Q;
    params[2 * C + c] = R;
  }
  __syncthreads();
  long i0 = ((long)blockIdx.x * NT + threadIdx.x) * 8;
  if (i0 >= total) return;
  if (i0 + 8 <= total) {  // vectorized b128 path (C % 8 == 0)
    int4 dv = *(const int4*)&dy[i0];
    int4 xv = *(const int4*)&x[i0];
    int4 yv;
    if (relu) yv = *(const int4*)&y[i0];
    const bf16_t* de = (const bf16_t*)&dv;
    const bf16_t* xe = (const bf16_t*)&xv;
    const bf16_t* ye = (const bf16_t*)&yv;
    __align__(16) bf16_t odx[8];
    __align__(16) bf16_t ores[8];
    int c0 = (int)(i0 % C);
    float4 p0 = *(const float4*)&params[c0];
    float4 p1 = *(const float4*)&params[c0 + 4];
    float4 q0 = *(const float4*)&params[C + c0];
    float4 q1 = *(const float4*)&params[C + c0 + 4];
    float4 r0 = *(const float4*)&params[2 * C + c0];
    float4 r1 = *(const float4*)&params[2 * C + c0 + 4];
    float pv[8] = {p0.x, p0.y, p0.z, p0.w, p1.x, p1.y, p1.z, p1.w};
    float qv[8] = {q0.x, q0.y, q0.z, q0.w, q1.x, q1.y, q1.z, q1.w};
    float rv[8] = {r0.x, r0.y, r0.z, r0.w, r1.x, r1.y, r1.z, r1.w};
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bf2f(de[j]);
      if (relu && bf2f(ye[j]) <= 0.f) g = 0.f;
      ores[j] = f2bf(g);
      float v = pv[j] * g + qv[j] * bf2f(xe[j]) + rv[j];
      odx[j] = f2bf(v);
    }
    *(int4*)&dx[i0] = *(int4*)odx;
    if (dres != nullptr) *(int4*)&dres[i0] = *(int4*)ores;
    return;
  }
  for (long i = i0; i < total; ++i) {
    int c = (int)(i % C);
    float g = bf2f(dy[i]);
    if (relu && bf2f(y[i]) <= 0.f) g = 0.f;
    if (dres != nullptr) dres[i] = f2bf(g);
    dx[i] = f2bf(params[c] * g + params[C + c] * bf2f(x[i])
                 + params[2 * C + c]);
  }
}

// ------------------------------------------------------------------- add+ReLU

__global__ __launch_bounds__(NT)
void add_relu_fwd_kernel(const bf16_t* __restrict__ a,
                         const bf16_t* __restrict__ b,
                         bf16_t* __restrict__ y, long total) {
  long i0 = ((long)blockIdx.x * NT + threadIdx.x) * 8;
  if (i0 >= total) return;
  if (i0 + 8 <= total) {
    int4 av = *(const int4*)&a[i0];
    int4 bv = *(const int4*)&b[i0];
    bf16_t* ae = (bf16_t*)&av;
    bf16_t* be = (bf16_t*)&bv;
    bf16_t out[8];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      out[j] = f2bf(fmaxf(bf2f(ae[j]) + bf2f(be[j]), 0.f));
    *(int4*)&y[i0] = *(int4*)out;
  } else {
    for (long i = i0; i < total; ++i)
      y[i] = f2bf(fmaxf(bf2f(a[i]) + bf2f(b[i]), 0.f));
  }
}

__global__ __launch_bounds__(NT)
void add_relu_bwd_kernel(const bf16_t* __restrict__ dy,
                         const bf16_t* __restrict__ y,
                         bf16_t* __restrict__ da, long total) {
  long i0 = ((long)blockIdx.x * NT + threadIdx.x) * 8;
  if (i0 >= total) return;
  long iend = min(i0 + 8, total);
  for (long i = i0; i < iend; ++i)
    da[i] = (bf2f(y[i]) > 0.f) ? dy[i] : (bf16_t)0;
}

// ---------------------------------------------------------------- DownsampleA
// y (N,H/2,W/2,2C): [:, :, :, :C] = x[:, ::2, ::2, :], rest zero.

__global__ __launch_bounds__(NT)
void downsample_a_fwd_kernel(const bf16_t* __restrict__ x,
                             bf16_t* __restrict__ y, int N, int H, int W,
                             int C, long total_out) {
  long i = (long)blockIdx.x * NT + threadIdx.x;
  if (i >= total_out) return;
  int C2 = 2 * C, Ho = H / 2, Wo = W / 2;
  int c = (int)(i % C2);
  long rest = i / C2;
  int wo = (int)(rest % Wo);
  rest /= Wo;
  int ho = (int)(rest % Ho);
  int n = (int)(rest / Ho);
  bf16_t v = 0;
  if (c < C)
    v = x[(((long)n * H + 2 * ho) * W + 2 * wo) * C + c];
  y[i] = v;
}

__global__ __launch_bounds__(NT)
void downsample_a_bwd_kernel(const bf16_t* __restrict__ dy,
                             bf16_t* __restrict__ dx, int N, int H, int W,
                             int C, long total_in) {
  long i = (long)blockIdx.x * NT + threadIdx.x;
  if (i >= total_in) return;
  int c = (int)(i % C);
  long rest = i / C;
  int w = (int)(rest % W);
  rest /= W;
  int h = (int)(rest % H);
  int n = (int)(rest / H);
  bf16_t v = 0;
  if ((h & 1) == 0 && (w & 1) == 0) {
    int Ho = H / 2, Wo = W / 2;
    v = dy[(((long)n * Ho + h / 2) * Wo + w / 2) * (2 * C) + c];
  }
  dx[i] = v;
}

// ------------------------------------------------------------ global avg pool

__global__ __launch_bounds__(NT)
void gap_fwd_kernel(const bf16_t* __restrict__ x, bf16_t* __restrict__ y,
                    int HW, int C) {
  const int n = blockIdx.x;
  const float scale = 1.f / (float)HW;
  for (int c = threadIdx.x; c < C; c += NT) {
    float s = 0.f;
    const bf16_t* base = x + (long)n * HW * C + c;
    for (int i = 0; i < HW; ++i) s += bf2f(base[(long)i * C]);
    y[(long)n * C + c] = f2bf(s * scale);
  }
}

__global__ __launch_bounds__(NT)
void gap_bwd_kernel(const bf16_t* __restrict__ dy, bf16_t* __restrict__ dx,
                    int HW, int C, long total) {
  long i = (long)blockIdx.x * NT + threadIdx.x;
  if (i >= total) return;
  int c = (int)(i % C);
  long n = i / ((long)HW * C);
  dx[i] = f2bf(bf2f(dy[n * C + c]) / (float)HW);
}

// ------------------------------------------------------------------- max pool

__global__ __launch_bounds__(NT)
void maxpool_fwd_kernel(const bf16_t* __restrict__ x, bf16_t* __restrict__ y,
                        int* __restrict__ idx, int N, int H, int W, int C,
                        int kk, int st, int pad, int Ho, int Wo,
                        long total_out) {
  long i = (long)blockIdx.x * NT + threadIdx.x;
  if (i >= total_out) return;
  int c = (int)(i % C);
  long rest = i / C;
  int wo = (int)(rest % Wo);
  rest /= Wo;
  int ho = (int)(rest % Ho);
  int n = (int)(rest / Ho);
  float best = -3.4e38f;
  int bi = 0;
  for (int r = 0; r < kk; ++r) {
    int hi = ho * st - pad + r;
    if (hi < 0 || hi >= H) continue;
    for (int s = 0; s < kk; ++s) {
      int wi = wo * st - pad + s;
      if (wi < 0 || wi >= W) continue;
      float v = bf2f(x[(((long)n * H + hi) * W + wi) * C + c]);
      if (v > best) { best = v; bi = hi * W + wi; }
    }
  }
  y[i] = f2bf(best);
  idx[i] = bi;
}

__global__ __launch_bounds__(NT)
void maxpool_bwd_kernel(const bf16_t* __restrict__ dy,
                        const int* __restrict__ idx, bf16_t* __restrict__ dx,
                        int N, int H, int W, int C, int kk, int st, int pad,
                        int Ho, int Wo, long total_in) {
  long i = (long)blockIdx.x * NT + threadIdx.x;
  if (i >= total_in) return;
  int c = (int)(i % C);
  long rest = i / C;
  int wi = (int)(rest % W);
  rest /= W;
  int hi = (int)(rest % H);
  int n = (int)(rest / H);
  int my = hi * W + wi;
  float acc = 0.f;
  // windows whose output could have selected (hi, wi)
  int ho_lo = max(0, (hi + pad - kk + st) / st), ho_hi = min(Ho - 1,
                                                            (hi + pad) / st);
  int wo_lo = max(0, (wi + pad - kk + st) / st), wo_hi = min(Wo - 1,
                                                            (wi + pad) / st);
  for (int ho = ho_lo; ho <= ho_hi; ++ho)
    for (int wo = wo_lo; wo <= wo_hi; ++wo) {
      long o = (((long)n * Ho + ho) * Wo + wo) * C + c;
      if (idx[o] == my) acc += bf2f(dy[o]);
    }
  dx[i] = f2bf(acc);
}

// 8-channel vectorized variant (C % 8 == 0): the scalar form moved ~2 B per
// thread and measured 280 us on the rn50 stem pool backward
__global__ __launch_bounds__(NT)
void maxpool_bwd_v_kernel(const bf16_t* __restrict__ dy,
                          const int* __restrict__ idx,
                          bf16_t* __restrict__ dx,
                          int N, int H, int W, int C, int kk, int st,
                          int pad, int Ho, int Wo, long total8) {
  long t8 = (long)blockIdx.x * NT + threadIdx.x;
  if (t8 >= total8) return;
  long i0 = t8 * 8;
  int c0 = (int)(i0 % C);
  long rest = i0 / C;
  int wi = (int)(rest % W);
  rest /= W;
  int hi = (int)(rest % H);
  int n = (int)(rest / H);
  int my = hi * W + wi;
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  int ho_lo = max(0, (hi + pad - kk + st) / st);
  int ho_hi = min(Ho - 1, (hi + pad) / st);
  int wo_lo = max(0, (wi + pad - kk + st) / st);
  int wo_hi = min(Wo - 1, (wi + pad) / st);
  for (int ho = ho_lo; ho <= ho_hi; ++ho)
    for (int wo = wo_lo; wo <= wo_hi; ++wo) {
      long o = (((long)n * Ho + ho) * Wo + wo) * C + c0;
      int4 iv0 = *(const int4*)&idx[o];
      int4 iv1 = *(const int4*)&idx[o + 4];
      int4 dv = *(const int4*)&dy[o];
      const int* ie0 = (const int*)&iv0;
      const int* ie1 = (const int*)&iv1;
      const bf16_t* de = (const bf16_t*)&dv;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        if (ie0[j] == my) acc[j] += bf2f(de[j]);
        if (ie1[j] == my) acc[4 + j] += bf2f(de[4 + j]);
      }
    }
  bf16_t out[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) out[j] = f2bf(acc[j]);
  *(int4*)&dx[i0] = *(int4*)out;
}

// ---------------------- strided 1x1 helpers (gather/scatter subsampled grid)

__global__ __launch_bounds__(NT)
void stride_gather_kernel(const bf16_t* __restrict__ x,
                          bf16_t* __restrict__ xg, int H, int W, int C,
                          int s, int Ho, int Wo, long total_out) {
  long i = (long)blockIdx.x * NT + threadIdx.x;
  if (i >= total_out) return;
  int c = (int)(i % C);
  long rest = i / C;
  int wo = (int)(rest % Wo);
  rest /= Wo;
  int ho = (int)(rest % Ho);
  int n = (int)(rest / Ho);
  xg[i] = x[(((long)n * H + ho * s) * W + wo * s) * C + c];
}

// 8-channel vectorized gather/scatter (C % 8 == 0)
__global__ __launch_bounds__(NT)
void stride_gather_v_kernel(const bf16_t* __restrict__ x,
                            bf16_t* __restrict__ xg, int H, int W, int C,
                            int s, int Ho, int Wo, long total8) {
  long t8 = (long)blockIdx.x * NT + threadIdx.x;
  if (t8 >= total8) return;
  long i0 = t8 * 8;
  int c = (int)(i0 % C);
  long rest = i0 / C;
  int wo = (int)(rest % Wo);
  rest /= Wo;
  int ho = (int)(rest % Ho);
  int n = (int)(rest / Ho);
  *(int4*)&xg[i0] =
      *(const int4*)&x[(((long)n * H + ho * s) * W + wo * s) * C + c];
}

__global__ __launch_bounds__(NT)
void stride_scatter_v_kernel(const bf16_t* __restrict__ dxs,
                             bf16_t* __restrict__ dx, int H, int W, int C,
                             int s, int Ho, int Wo, long total8) {
  long t8 = (long)blockIdx.x * NT + threadIdx.x;
  if (t8 >= total8) return;
  long i0 = t8 * 8;
  int c = (int)(i0 % C);
  long rest = i0 / C;
  int w = (int)(rest % W);
  rest /= W;
  int h = (int)(rest % H);
  int n = (int)(rest / H);
  int4 v{0, 0, 0, 0};
  if (h % s == 0 && w % s == 0) {
    int ho = h / s, wo = w / s;
    if (ho < Ho && wo < Wo)
      v = *(const int4*)&dxs[(((long)n * Ho + ho) * Wo + wo) * C + c];
  }
  *(int4*)&dx[i0] = v;
}

__global__ __launch_bounds__(NT)
void stride_scatter_kernel(const bf16_t* __restrict__ dxs,
                           bf16_t* __restrict__ dx, int H, int W, int C,
                           int s, int Ho, int Wo, long total_in) {
  long i = (long)blockIdx.x * NT + threadIdx.x;
  if (i >= total_in) return;
  int c = (int)(i % C);
  long rest = i / C;
  int w = (int)(rest % W);
  rest /= W;
  int h = (int)(rest % H);
  int n = (int)(rest / H);
  bf16_t v = 0;
  if (h % s == 0 && w % s == 0) {
    int ho = h / s, wo = w / s;
    if (ho < Ho && wo < Wo)
      v = dxs[(((long)n * Ho + ho) * Wo + wo) * C + c];
  }
  dx[i] = v;
}

// ============================== launchers ==============================

extern "C" {

void cilfw_bn_fwd(const void* x, void* y, const void* res,
                  const void* gamma, const void* beta,
                  void* running_mean, void* running_var, void* mean,
                  void* invstd, void* scratch_sums, const void* ext_part,
                  int ext_gy, long M, int C,
                  float momentum, float eps, int training, int relu,
                  void* stream) {
  hipStream_t st = (hipStream_t)stream;
  if (training && ext_part != nullptr) {
    // partial sums already produced by the PRODUCING conv's epilogue
    // (conv2d_fwd_v2 bn_parts) — one fused reduce+finalize launch replaces
    // the bn_sums pass (and its full re-read of x)
    hipLaunchKernelGGL(bn_reduce_finalize_kernel,
                       dim3(cdiv(C, NT / WAVE)), dim3(NT), 0, st,
                       (const float*)ext_part, (float*)mean, (float*)invstd,
                       (float*)running_mean, (float*)running_var, ext_gy, M,
                       C, momentum, eps);
  } else if (training) {
    // scratch_sums: [gy][2][C] partials + [2][C] reduced (see wrapper sizing)
    int rows_per_blk = 256;
    dim3 grid(cdiv(C, 64), cdiv((int)min(M, (long)INT32_MAX), rows_per_blk));
    float* part = (float*)scratch_sums;
    float* sum = part + (long)grid.y * 2 * C;
    float* sumsq = sum + C;
    hipLaunchKernelGGL(bn_sums_kernel, grid, dim3(NT), 0, st,
                       (const bf16_t*)x, part, M, C, rows_per_blk);
    hipLaunchKernelGGL(bn_reduce_slabs_kernel,
                       dim3(cdiv(2 * C, NT / WAVE)), dim3(NT), 0, st, part,
                       sum, (int)grid.y, (long)2 * C, (float*)nullptr,
                       (long)2 * C);
    hipLaunchKernelGGL(bn_finalize_kernel, dim3(cdiv(C, 256)), dim3(256), 0,
                       st, sum, sumsq, (float*)mean, (float*)invstd,
                       (float*)running_mean, (float*)running_var, M, C,
                       momentum, eps, 1);
  } else {
    hipLaunchKernelGGL(bn_finalize_kernel, dim3(cdiv(C, 256)), dim3(256), 0,
                       st, nullptr, nullptr, (float*)mean, (float*)invstd,
                       (float*)running_mean, (float*)running_var, M, C,
                       momentum, eps, 0);
  }
  long total = M * C;
  long blocks = cdiv((long)total, (long)NT * 8);
  hipLaunchKernelGGL(bn_apply_kernel, dim3((int)blocks), dim3(NT),
                     2 * C * sizeof(float), st, (const bf16_t*)x, (bf16_t*)y,
                     (const bf16_t*)res, (const float*)gamma,
                     (const float*)beta, (const float*)mean,
                     (const float*)invstd, total, C, relu);
}

void cilfw_bn_apply_only(const void* x, void* y, const void* res,
                         const void* gamma, const void* beta,
                         const void* mean, const void* invstd, long total,
                         int C, int relu, void* stream) {
  // frozen-model eval: stats precomputed, apply is the only launch
  long blocks = cdiv((long)total, (long)NT * 8);
  hipLaunchKernelGGL(bn_apply_kernel, dim3((int)blocks), dim3(NT),
                     2 * C * sizeof(float), (hipStream_t)stream,
                     (const bf16_t*)x, (bf16_t*)y, (const bf16_t*)res,
                     (const float*)gamma, (const float*)beta,
                     (const float*)mean, (const float*)invstd, total, C,
                     relu);
}

void cilfw_bn_bwd(const void* dy, const void* x, const void* y, void* dx,
                  void* dres, const void* gamma, const void* mean,
                  const void* invstd, void* dgb, void* dg_out, void* db_out,
                  const void* ext_part, int ext_gy,
                  long M, int C, int relu, int training, void* stream) {
  // dgb: [gy][2][C] partials followed by the reduced [dgamma | dbeta];
  // dg_out/db_out (optional) divert the reduced grads into flat-grad slots.
  // ext_part/ext_gy: [gy][2][C] (dgamma, dbeta) partials ALREADY produced by
  // the upstream conv's bwd-data epilogue (this dy is that conv's dx) — the
  // sums pass and its 3-stream re-read of (dy, x, y) are skipped; dgb then
  // only needs room for the reduced [dgamma | dbeta].
  hipStream_t st = (hipStream_t)stream;
  int rows_per_blk = 256;
  dim3 grid(cdiv(C, 64), cdiv((int)min(M, (long)INT32_MAX), rows_per_blk));
  int gy = ext_part ? ext_gy : (int)grid.y;
  long red_off = ext_part ? 0 : (long)grid.y * 2 * C;
  float* part = ext_part ? (float*)ext_part : (float*)dgb;
  float* dgamma = dg_out ? (float*)dg_out : (float*)dgb + red_off;
  float* dbeta = db_out ? (float*)db_out : (float*)dgb + red_off + C;
  if (ext_part == nullptr)
    hipLaunchKernelGGL(bn_bwd_sums_kernel, grid, dim3(NT), 0, st,
                       (const bf16_t*)dy, (const bf16_t*)x, (const bf16_t*)y,
                       (const float*)mean, (const float*)invstd, part,
                       M, C, rows_per_blk, relu);
  hipLaunchKernelGGL(bn_reduce_slabs_kernel, dim3(cdiv(2 * C, NT / WAVE)),
                     dim3(NT), 0, st, part, dgamma, gy,
                     (long)2 * C, dbeta, (long)C);
  long total = M * C;
  long blocks = cdiv((long)total, (long)NT * 8);
  hipLaunchKernelGGL(bn_bwd_apply_kernel, dim3((int)blocks), dim3(NT),
                     3 * C * sizeof(float), st, (const bf16_t*)dy,
                     (const bf16_t*)x, (const bf16_t*)y, (bf16_t*)dx,
                     (bf16_t*)dres, (const float*)gamma, (const float*)mean,
                     (const float*)invstd, (const float*)dgamma,
                     (const float*)dbeta, total, M, C, relu, training);
}

void cilfw_add_relu_fwd(const void* a, const void* b, void* y, long total,
                        void* stream) {
  long blocks = cdiv((long)total, (long)NT * 8);
  hipLaunchKernelGGL(add_relu_fwd_kernel, dim3((int)blocks), dim3(NT), 0,
                     (hipStream_t)stream, (const bf16_t*)a, (const bf16_t*)b,
                     (bf16_t*)y, total);
}

void cilfw_add_relu_bwd(const void* dy, const void* y, void* da, long total,
                        void* stream) {
  long blocks = cdiv((long)total, (long)NT * 8);
  hipLaunchKernelGGL(add_relu_bwd_kernel, dim3((int)blocks), dim3(NT), 0,
                     (hipStream_t)stream, (const bf16_t*)dy, (const bf16_t*)y,
                     (bf16_t*)da, total);
}

void cilfw_downsample_a_fwd(const void* x, void* y, int N, int H, int W,
                            int C, void* stream) {
  long total = (long)N * (H / 2) * (W / 2) * 2 * C;
  hipLaunchKernelGGL(downsample_a_fwd_kernel,
                     dim3((int)cdiv((long)total, (long)NT)), dim3(NT), 0,
                     (hipStream_t)stream, (const bf16_t*)x, (bf16_t*)y, N, H,
                     W, C, total);
}

void cilfw_downsample_a_bwd(const void* dy, void* dx, int N, int H, int W,
                            int C, void* stream) {
  long total = (long)N * H * W * C;
  hipLaunchKernelGGL(downsample_a_bwd_kernel,
                     dim3((int)cdiv((long)total, (long)NT)), dim3(NT), 0,
                     (hipStream_t)stream, (const bf16_t*)dy, (bf16_t*)dx, N,
                     H, W, C, total);
}

void cilfw_gap_fwd(const void* x, void* y, int N, int HW, int C,
                   void* stream) {
  hipLaunchKernelGGL(gap_fwd_kernel, dim3(N), dim3(NT), 0,
                     (hipStream_t)stream, (const bf16_t*)x, (bf16_t*)y, HW, C);
}

void cilfw_gap_bwd(const void* dy, void* dx, int N, int HW, int C,
                   void* stream) {
  long total = (long)N * HW * C;
  hipLaunchKernelGGL(gap_bwd_kernel, dim3((int)cdiv((long)total, (long)NT)),
                     dim3(NT), 0, (hipStream_t)stream, (const bf16_t*)dy,
                     (bf16_t*)dx, HW, C, total);
}

void cilfw_maxpool_fwd(const void* x, void* y, void* idx, int N, int H, int W,
                       int C, int kk, int st, int pad, int Ho, int Wo,
                       void* stream) {
  long total = (long)N * Ho * Wo * C;
  hipLaunchKernelGGL(maxpool_fwd_kernel,
                     dim3((int)cdiv((long)total, (long)NT)), dim3(NT), 0,
                     (hipStream_t)stream, (const bf16_t*)x, (bf16_t*)y,
                     (int*)idx, N, H, W, C, kk, st, pad, Ho, Wo, total);
}

void cilfw_maxpool_bwd(const void* dy, const void* idx, void* dx, int N,
                       int H, int W, int C, int kk, int st, int pad, int Ho,
                       int Wo, void* stream) {
  long total = (long)N * H * W * C;
  if (C % 8 == 0) {
    long t8 = total / 8;
    hipLaunchKernelGGL(maxpool_bwd_v_kernel,
                       dim3((int)cdiv((long)t8, (long)NT)), dim3(NT), 0,
                       (hipStream_t)stream, (const bf16_t*)dy,
                       (const int*)idx, (bf16_t*)dx, N, H, W, C, kk, st,
                       pad, Ho, Wo, t8);
  } else {
    hipLaunchKernelGGL(maxpool_bwd_kernel,
                       dim3((int)cdiv((long)total, (long)NT)), dim3(NT), 0,
                       (hipStream_t)stream, (const bf16_t*)dy,
                       (const int*)idx, (bf16_t*)dx, N, H, W, C, kk, st,
                       pad, Ho, Wo, total);
  }
}

void cilfw_stride_gather(const void* x, void* xg, int N, int H, int W,
                         int C, int s, int Ho, int Wo, void* stream) {
  long total = (long)N * Ho * Wo * C;
  if (C % 8 == 0) {
    long t8 = total / 8;
    hipLaunchKernelGGL(stride_gather_v_kernel,
                       dim3((int)cdiv((long)t8, (long)NT)), dim3(NT), 0,
                       (hipStream_t)stream, (const bf16_t*)x, (bf16_t*)xg,
                       H, W, C, s, Ho, Wo, t8);
  } else {
    hipLaunchKernelGGL(stride_gather_kernel,
                       dim3((int)cdiv((long)total, (long)NT)), dim3(NT), 0,
                       (hipStream_t)stream, (const bf16_t*)x, (bf16_t*)xg,
                       H, W, C, s, Ho, Wo, total);
  }
}

void cilfw_stride_scatter(const void* dxs, void* dx, int N, int H, int W,
                          int C, int s, int Ho, int Wo, void* stream) {
  long total = (long)N * H * W * C;
  if (C % 8 == 0) {
    long t8 = total / 8;
    hipLaunchKernelGGL(stride_scatter_v_kernel,
                       dim3((int)cdiv((long)t8, (long)NT)), dim3(NT), 0,
                       (hipStream_t)stream, (const bf16_t*)dxs, (bf16_t*)dx,
                       H, W, C, s, Ho, Wo, t8);
  } else {
    hipLaunchKernelGGL(stride_scatter_kernel,
                       dim3((int)cdiv((long)total, (long)NT)), dim3(NT), 0,
                       (hipStream_t)stream, (const bf16_t*)dxs, (bf16_t*)dx,
                       H, W, C, s, Ho, Wo, total);
  }
}

}  // extern "C"
