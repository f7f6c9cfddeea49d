// cilfw — implicit-GEMM MFMA convolutions for gfx950 (CDNA4), NHWC bf16.
//
// Replaces the reference's cuDNN conv kernels.
// Formulation: Y[M=N*Ho*Wo, K] = im2col(X)[M, C*R*S] @ W[C*R*S, K], W stored
// (R,S,C,K) so the B operand is dense with k-index (r*S+s)*C+c.
//
// Tile: 128(M) x 64(Kout) x 32(K-step), 4 waves as 2x2, each wave 64x32 via
// mfma_f32_16x16x32_bf16 (M_rep=4, N_rep=2), fp32 accumulators, double-buffered
// LDS with +16B row padding (bank-conflict-free ds_read_b128).
// Fast A-staging path when C % 32 == 0 (every K-chunk lies inside one (r,s) —
// 16B vector loads); generic per-element gather otherwise (stem convs).

#include "common.h"

#define BM 128
#define BN 64
#define BK 32
#define LP (BK + 8)          // LDS row pitch in bf16 elements (+16B pad)
#define NTHREADS 256

// one __shared__ object only (see guide §5 trap 4a)
// As: 2 * BM * LP, Bs: 2 * BN * LP
#define AS_OFF(buf) ((buf) * BM * LP)
#define BS_OFF(buf) (2 * BM * LP + (buf) * BN * LP)
#define LDS_ELEMS (2 * BM * LP + 2 * BN * LP)

struct ConvGeom {
  int N, H, W, C, K, R, S, stride, pad, Ho, Wo;
};

// ---- shared MFMA core: given staged As/Bs, accumulate 4x2 fragments ----
DEV void mfma_tile(const bf16_t* lds, int a_off, int b_off, int wr, int wc,
                   f32x4 acc[4][2]) {
  FragIdx fi = frag_idx();
  const int kb = fi.quad * 8;  // this lane's k-offset within the 32-chunk
#pragma unroll
  for (int mr = 0; mr < 4; ++mr) {
    int row = wr * 64 + mr * 16 + fi.half;
    bf16x8 a = *(const bf16x8*)&lds[a_off + row * LP + kb];
#pragma unroll
    for (int nr = 0; nr < 2; ++nr) {
      int col = wc * 32 + nr * 16 + fi.half;
      bf16x8 b = *(const bf16x8*)&lds[b_off + col * LP + kb];
      acc[mr][nr] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[mr][nr],
                                                            0, 0, 0);
    }
  }
}

// ============================== forward ==============================

__global__ __launch_bounds__(NTHREADS)
void conv2d_fwd_kernel(const bf16_t* __restrict__ x,
                       const bf16_t* __restrict__ w,
                       bf16_t* __restrict__ y, ConvGeom g, int M, int CRS,
                       int nk, int fast_a) {
  __shared__ bf16_t lds[LDS_ELEMS];
  const int m0 = blockIdx.x * BM;
  const int ko0 = blockIdx.y * BN;
  const int t = threadIdx.x;
  const int wave = t >> 6, wr = wave >> 1, wc = wave & 1;

  // per-thread A-staging coords: 2 threads per row, 16 elems each
  const int arow = t >> 1, ahalf = t & 1;
  int m = m0 + arow;
  int an = 0, aho = 0, awo = 0;
  bool arow_ok = m < M;
  if (arow_ok) {
    an = m / (g.Ho * g.Wo);
    int rem = m - an * (g.Ho * g.Wo);
    aho = rem / g.Wo;
    awo = rem - aho * g.Wo;
  }
  const int ahb = aho * g.stride - g.pad;  // base input coords
  const int awb = awo * g.stride - g.pad;

  f32x4 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0, 0, 0, 0};

  // staging registers
  int4 areg0, areg1;
  bf16_t breg[8];

  auto stage_to_regs = [&](int kt) {
    const int k0 = kt * BK;
    // ---- A: 16 elems for (arow, ahalf) ----
    if (fast_a) {
      int k = k0 + ahalf * 16;
      int rs = k / g.C, c0 = k - rs * g.C;
      int r = rs / g.S, s = rs - r * g.S;
      int hi = ahb + r, wi = awb + s;
      if (arow_ok && hi >= 0 && hi < g.H && wi >= 0 && wi < g.W) {
        const int4* src = (const int4*)&x[(((long)an * g.H + hi) * g.W + wi)
                                          * g.C + c0];
        areg0 = src[0];
        areg1 = src[1];
      } else {
        areg0 = int4{0, 0, 0, 0};
        areg1 = int4{0, 0, 0, 0};
      }
    } else {
      bf16_t tmp[16];
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        int k = k0 + ahalf * 16 + j;
        bf16_t v = 0;
        if (arow_ok && k < CRS) {
          int rs = k / g.C, c = k - rs * g.C;
          int r = rs / g.S, s = rs - r * g.S;
          int hi = ahb + r, wi = awb + s;
          if (hi >= 0 && hi < g.H && wi >= 0 && wi < g.W)
            v = x[(((long)an * g.H + hi) * g.W + wi) * g.C + c];
        }
        tmp[j] = v;
      }
      areg0 = *(int4*)&tmp[0];
      areg1 = *(int4*)&tmp[8];
    }
    // ---- B: Bs[n][kk] = w[(k0+kk)*K + ko0+n]
    // thread: kk = t>>3 (0..31), ng = t&7 (8 couts per b128 load; K % 8 == 0)
    const int bkk = t >> 3, bng = t & 7;
    {
      int k = k0 + bkk;
      if (k < CRS && ko0 + bng * 8 + 8 <= g.K) {
        *(int4*)breg = *(const int4*)&w[(long)k * g.K + ko0 + bng * 8];
      } else if (k < CRS && ko0 + bng * 8 < g.K) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          breg[j] = (ko0 + bng * 8 + j < g.K)
                        ? w[(long)k * g.K + ko0 + bng * 8 + j] : 0;
      } else {
        *(int4*)breg = int4{0, 0, 0, 0};
      }
    }
  };

  auto regs_to_lds = [&](int buf) {
    bf16_t* As = &lds[AS_OFF(buf)];
    bf16_t* Bs = &lds[BS_OFF(buf)];
    *(int4*)&As[arow * LP + ahalf * 16] = areg0;
    *(int4*)&As[arow * LP + ahalf * 16 + 8] = areg1;
    const int bkk = t >> 3, bng = t & 7;
#pragma unroll
    for (int j = 0; j < 8; ++j) Bs[(bng * 8 + j) * LP + bkk] = breg[j];
  };

  stage_to_regs(0);
  regs_to_lds(0);
  __syncthreads();
  for (int kt = 0; kt < nk; ++kt) {
    int cur = kt & 1;
    if (kt + 1 < nk) stage_to_regs(kt + 1);      // global loads overlap MFMA
    mfma_tile(lds, AS_OFF(cur), BS_OFF(cur), wr, wc, acc);
    __syncthreads();
    if (kt + 1 < nk) {
      regs_to_lds(cur ^ 1);
      __syncthreads();
    }
  }

  // epilogue: C/D map col=lane&15, row=(lane>>4)*4+reg
  FragIdx fi = frag_idx();
#pragma unroll
  for (int mr = 0; mr < 4; ++mr)
#pragma unroll
    for (int nr = 0; nr < 2; ++nr)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr * 64 + mr * 16 + fi.quad * 4 + r;
        int col = ko0 + wc * 32 + nr * 16 + fi.half;
        if (row < M && col < g.K)
          y[(long)row * g.K + col] = f2bf(acc[mr][nr][r]);
      }
}

// ============================== backward data ==============================
// dX[M=N*H*W, C] = gather(dY)[M, R*S*K] @ B where B[(r,s,k)][c] = W[r,s,c,k].

__global__ __launch_bounds__(NTHREADS)
void conv2d_bwd_data_kernel(const bf16_t* __restrict__ dy,
                            const bf16_t* __restrict__ w,
                            bf16_t* __restrict__ dx, ConvGeom g, int M,
                            int RSK, int nk, int fast_a) {
  __shared__ bf16_t lds[LDS_ELEMS];
  const int m0 = blockIdx.x * BM;
  const int c0 = blockIdx.y * BN;
  const int t = threadIdx.x;
  const int wave = t >> 6, wr = wave >> 1, wc = wave & 1;

  const int arow = t >> 1, ahalf = t & 1;
  int m = m0 + arow;
  int an = 0, ahi = 0, awi = 0;
  bool arow_ok = m < M;
  if (arow_ok) {
    an = m / (g.H * g.W);
    int rem = m - an * (g.H * g.W);
    ahi = rem / g.W;
    awi = rem - ahi * g.W;
  }

  f32x4 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0, 0, 0, 0};

  int4 areg0, areg1;
  bf16_t breg[8];

  auto stage_to_regs = [&](int kt) {
    const int k0 = kt * BK;
    if (fast_a) {  // K % 32 == 0: chunk inside one (r,s)
      int k = k0 + ahalf * 16;
      int rs = k / g.K, kc0 = k - rs * g.K;
      int r = rs / g.S, s = rs - r * g.S;
      int ho2 = ahi + g.pad - r, wo2 = awi + g.pad - s;
      bool ok = arow_ok && ho2 >= 0 && wo2 >= 0 &&
                (ho2 % g.stride) == 0 && (wo2 % g.stride) == 0;
      int ho = ho2 / g.stride, wo = wo2 / g.stride;
      ok = ok && ho < g.Ho && wo < g.Wo;
      if (ok) {
        const int4* src = (const int4*)&dy[(((long)an * g.Ho + ho) * g.Wo + wo)
                                           * g.K + kc0];
        areg0 = src[0];
        areg1 = src[1];
      } else {
        areg0 = int4{0, 0, 0, 0};
        areg1 = int4{0, 0, 0, 0};
      }
    } else {
      bf16_t tmp[16];
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        int k = k0 + ahalf * 16 + j;
        bf16_t v = 0;
        if (arow_ok && k < RSK) {
          int rs = k / g.K, kc = k - rs * g.K;
          int r = rs / g.S, s = rs - r * g.S;
          int ho2 = ahi + g.pad - r, wo2 = awi + g.pad - s;
          if (ho2 >= 0 && wo2 >= 0 && (ho2 % g.stride) == 0 &&
              (wo2 % g.stride) == 0) {
            int ho = ho2 / g.stride, wo = wo2 / g.stride;
            if (ho < g.Ho && wo < g.Wo)
              v = dy[(((long)an * g.Ho + ho) * g.Wo + wo) * g.K + kc];
          }
        }
        tmp[j] = v;
      }
      areg0 = *(int4*)&tmp[0];
      areg1 = *(int4*)&tmp[8];
    }
    // B: Bs[c][kk] = w[(rs*C + c0+c)*K + kc]
    // thread: c = t>>2 (0..63), slot = t&3 -> 8 contiguous kc (K % 8 == 0 so
    // an 8-chunk never crosses an (r,s) boundary)
    const int bc = t >> 2, bkk = (t & 3) * 8;
    {
      int k = k0 + bkk;
      if (k < RSK && c0 + bc < g.C) {
        int rs = k / g.K, kc = k - rs * g.K;
        *(int4*)breg = *(const int4*)&w[((long)rs * g.C + c0 + bc) * g.K + kc];
      } else {
        *(int4*)breg = int4{0, 0, 0, 0};
      }
    }
  };

  auto regs_to_lds = [&](int buf) {
    bf16_t* As = &lds[AS_OFF(buf)];
    bf16_t* Bs = &lds[BS_OFF(buf)];
    *(int4*)&As[arow * LP + ahalf * 16] = areg0;
    *(int4*)&As[arow * LP + ahalf * 16 + 8] = areg1;
    const int bc = t >> 2, bkk = (t & 3) * 8;
    *(int4*)&Bs[bc * LP + bkk] = *(int4*)breg;
  };

  stage_to_regs(0);
  regs_to_lds(0);
  __syncthreads();
  for (int kt = 0; kt < nk; ++kt) {
    int cur = kt & 1;
    if (kt + 1 < nk) stage_to_regs(kt + 1);
    mfma_tile(lds, AS_OFF(cur), BS_OFF(cur), wr, wc, acc);
    __syncthreads();
    if (kt + 1 < nk) {
      regs_to_lds(cur ^ 1);
      __syncthreads();
    }
  }

  FragIdx fi = frag_idx();
#pragma unroll
  for (int mr = 0; mr < 4; ++mr)
#pragma unroll
    for (int nr = 0; nr < 2; ++nr)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr * 64 + mr * 16 + fi.quad * 4 + r;
        int col = c0 + wc * 32 + nr * 16 + fi.half;
        if (row < M && col < g.C)
          dx[(long)row * g.C + col] = f2bf(acc[mr][nr][r]);
      }
}

// ============================== backward weight ==============================
// dW[(r,s,c), k] += sum_m X[m -> (n,hi,wi,c)] * dY[m, k], split over m-slices
// with fp32 atomics (dW pre-zeroed). Tile 64(CRS) x 64(K) x 32(m), waves 2x2.

#define WBM 64
#define WLDS_ELEMS (2 * WBM * LP + 2 * BN * LP)
#define WAS_OFF(buf) ((buf) * WBM * LP)
#define WBS_OFF(buf) (2 * WBM * LP + (buf) * BN * LP)

__global__ __launch_bounds__(NTHREADS)
void conv2d_bwd_weight_kernel(const bf16_t* __restrict__ dy,
                              const bf16_t* __restrict__ x,
                              float* __restrict__ dw, ConvGeom g, int M,
                              int CRS, int slice_len, int fast_a) {
  __shared__ bf16_t lds[WLDS_ELEMS];
  const int rs0 = blockIdx.x * WBM;   // CRS rows
  const int ko0 = blockIdx.y * BN;    // Kout cols
  const int ms = blockIdx.z * slice_len;
  const int me = min(ms + slice_len, M);
  const int t = threadIdx.x;
  const int wave = t >> 6, wr = wave >> 1, wc = wave & 1;

  // staging mapping: mm = t>>3 (0..31 reduction cols), grp = t&7 (8-row group)
  const int amm = t >> 3, agrp = t & 7;
  // A rows rs0 + agrp*8 + j (j=0..7): when C % 8 == 0 the 8-row group stays in
  // ONE (r,s) with contiguous c -> one b128 gather per m (fast path)
  int r_ = 0, s_ = 0, cbase_ = 0;
  const int rowb = rs0 + agrp * 8;
  bool agrp_ok = rowb < CRS;
  if (agrp_ok) {
    int rs = rowb / g.C;
    cbase_ = rowb - rs * g.C;
    r_ = rs / g.S;
    s_ = rs - r_ * g.S;
  }

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0, 0, 0, 0};

  __align__(16) bf16_t areg[8];
  __align__(16) bf16_t breg[8];
  const int HoWo = g.Ho * g.Wo;

  auto stage_to_regs = [&](int m0) {
    const int m = m0 + amm;
    int n = 0, ho = 0, wo = 0;
    const bool m_ok = m < me;
    if (m_ok) {
      n = m / HoWo;
      int rem = m - n * HoWo;
      ho = rem / g.Wo;
      wo = rem - ho * g.Wo;
    }
    // ---- A: 8 rows (rsc) x 1 col (m): b128 gather from x
    if (fast_a) {
      int hi = ho * g.stride - g.pad + r_;
      int wi = wo * g.stride - g.pad + s_;
      if (agrp_ok && m_ok && hi >= 0 && hi < g.H && wi >= 0 && wi < g.W) {
        *(int4*)areg = *(const int4*)&x[(((long)n * g.H + hi) * g.W + wi)
                                        * g.C + cbase_];
      } else {
        *(int4*)areg = int4{0, 0, 0, 0};
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int row = rowb + j;
        bf16_t v = 0;
        if (row < CRS && m_ok) {
          int rs = row / g.C, c = row - rs * g.C;
          int r = rs / g.S, s = rs - r * g.S;
          int hi = ho * g.stride - g.pad + r;
          int wi = wo * g.stride - g.pad + s;
          if (hi >= 0 && hi < g.H && wi >= 0 && wi < g.W)
            v = x[(((long)n * g.H + hi) * g.W + wi) * g.C + c];
        }
        areg[j] = v;
      }
    }
    // ---- B: 8 kouts x 1 col (m): b128 from dy (K % 8 == 0)
    if (m_ok && ko0 + agrp * 8 + 8 <= g.K) {
      *(int4*)breg = *(const int4*)&dy[(long)m * g.K + ko0 + agrp * 8];
    } else if (m_ok) {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        breg[j] = (ko0 + agrp * 8 + j < g.K)
                      ? dy[(long)m * g.K + ko0 + agrp * 8 + j] : 0;
    } else {
      *(int4*)breg = int4{0, 0, 0, 0};
    }
  };

  auto regs_to_lds = [&](int buf) {
    bf16_t* As = &lds[WAS_OFF(buf)];
    bf16_t* Bs = &lds[WBS_OFF(buf)];
#pragma unroll
    for (int j = 0; j < 8; ++j) As[(agrp * 8 + j) * LP + amm] = areg[j];
#pragma unroll
    for (int j = 0; j < 8; ++j) Bs[(agrp * 8 + j) * LP + amm] = breg[j];
  };

  const int nk = cdiv_i(me - ms, BK);
  stage_to_regs(ms);
  regs_to_lds(0);
  __syncthreads();
  for (int kt = 0; kt < nk; ++kt) {
    int cur = kt & 1;
    if (kt + 1 < nk) stage_to_regs(ms + (kt + 1) * BK);
    // wave tile 32x32: M_rep=2, N_rep=2
    {
      FragIdx fi = frag_idx();
      const int kb = fi.quad * 8;
      const bf16_t* As = &lds[WAS_OFF(cur)];
      const bf16_t* Bs = &lds[WBS_OFF(cur)];
#pragma unroll
      for (int mr = 0; mr < 2; ++mr) {
        int row = wr * 32 + mr * 16 + fi.half;
        bf16x8 a = *(const bf16x8*)&As[row * LP + kb];
#pragma unroll
        for (int nr = 0; nr < 2; ++nr) {
          int col = wc * 32 + nr * 16 + fi.half;
          bf16x8 b = *(const bf16x8*)&Bs[col * LP + kb];
          acc[mr][nr] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b,
                                                                acc[mr][nr],
                                                                0, 0, 0);
        }
      }
    }
    __syncthreads();
    if (kt + 1 < nk) {
      regs_to_lds(cur ^ 1);
      __syncthreads();
    }
  }

  FragIdx fi = frag_idx();
#pragma unroll
  for (int mr = 0; mr < 2; ++mr)
#pragma unroll
    for (int nr = 0; nr < 2; ++nr)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = rs0 + wr * 32 + mr * 16 + fi.quad * 4 + r;
        int col = ko0 + wc * 32 + nr * 16 + fi.half;
        if (row < CRS && col < g.K)
          atomicAdd(&dw[(long)row * g.K + col], acc[mr][nr][r]);
      }
}

// ============================== launchers ==============================

extern "C" {

void cilfw_conv2d_fwd(const void* x, const void* w, void* y,
                      int N, int H, int W, int C, int K, int R, int S,
                      int stride, int pad, int Ho, int Wo, void* stream) {
  ConvGeom g{N, H, W, C, K, R, S, stride, pad, Ho, Wo};
  int M = N * Ho * Wo;
  int CRS = C * R * S;
  int nk = cdiv(CRS, BK);
  int fast_a = (C % BK == 0);
  dim3 grid(cdiv(M, BM), cdiv(K, BN));
  hipLaunchKernelGGL(conv2d_fwd_kernel, grid, dim3(NTHREADS), 0,
                     (hipStream_t)stream, (const bf16_t*)x, (const bf16_t*)w,
                     (bf16_t*)y, g, M, CRS, nk, fast_a);
}

void cilfw_conv2d_bwd_data(const void* dy, const void* w, void* dx,
                           int N, int H, int W, int C, int K, int R, int S,
                           int stride, int pad, int Ho, int Wo, void* stream) {
  ConvGeom g{N, H, W, C, K, R, S, stride, pad, Ho, Wo};
  int M = N * H * W;
  int RSK = R * S * K;
  int nk = cdiv(RSK, BK);
  int fast_a = (K % BK == 0);
  dim3 grid(cdiv(M, BM), cdiv(C, BN));
  hipLaunchKernelGGL(conv2d_bwd_data_kernel, grid, dim3(NTHREADS), 0,
                     (hipStream_t)stream, (const bf16_t*)dy, (const bf16_t*)w,
                     (bf16_t*)dx, g, M, RSK, nk, fast_a);
}

void cilfw_conv2d_bwd_weight(const void* dy, const void* x, void* dw,
                             int N, int H, int W, int C, int K, int R, int S,
                             int stride, int pad, int Ho, int Wo,
                             void* stream) {
  ConvGeom g{N, H, W, C, K, R, S, stride, pad, Ho, Wo};
  int M = N * Ho * Wo;
  int CRS = C * R * S;
  (void)hipMemsetAsync(dw, 0, (size_t)CRS * K * sizeof(float),
                       (hipStream_t)stream);
  int slice_len = 4096;
  int nslices = cdiv(M, slice_len);
  int fast_a = (C % 8 == 0);
  dim3 grid(cdiv(CRS, WBM), cdiv(K, BN), nslices);
  hipLaunchKernelGGL(conv2d_bwd_weight_kernel, grid, dim3(NTHREADS), 0,
                     (hipStream_t)stream, (const bf16_t*)dy, (const bf16_t*)x,
                     (float*)dw, g, M, CRS, slice_len, fast_a);
}

}  // extern "C"
