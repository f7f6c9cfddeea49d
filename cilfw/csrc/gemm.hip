// cilfw — small MFMA GEMMs for the growable classifier head (fused single-GEMM
// forward over concatenated head weights — SURVEY.md §2.3 K7 — plus backward).
// Shapes are tiny (batch x feat x classes <= 128 x 2048 x 1000), so this favors
// a simple, fully bounds-checked 64x64x32 tile with generic staging.
//
//   linear_fwd : Y[M,N]  = X[M,K] @ W[N,K]^T (+ bias)   (bf16 out)
//   linear_dx  : dX[M,K] = dY[M,N] @ W[N,K]             (bf16 out)
//   linear_dw  : dW[N,K] = dY[M,N]^T @ X[M,K] (+ db)    (fp32 out)

#include "common.h"

#define GBM 64
#define GBN 64
#define GBK 32
#define GLP (GBK + 8)
#define GNT 256

// staged tiles: At[GBM][GLP], Bt[GBN][GLP]; fragment reads need
// At[row][k0..k0+7], Bt[col][k0..k0+7] contiguous.
#define GAS_OFF(buf) ((buf) * GBM * GLP)
#define GBS_OFF(buf) (2 * GBM * GLP + (buf) * GBN * GLP)
#define GLDS_ELEMS (2 * GBM * GLP + 2 * GBN * GLP)

// slab reduces live in conv.hip (same .so)
__global__ void reduce_slabs_bf16_kernel(const float*, bf16_t*, int, long);
__global__ void reduce_slabs_f32_kernel(const float*, float*, int, long, int);

// MODE: 0 = NT (fwd), 1 = NN (dx), 2 = TN (dw)
// ksplit: the head shapes are tiny in M,N but long in K (64 x 10..1000 x
// 2048 -> ONE 64x64 block, 166 us measured on rn50) — blockIdx.z slices
// the K loop into fp32 slabs reduced by the conv slab kernels.
template <int MODE, bool OUT_F32>
__global__ __launch_bounds__(GNT)
void gemm_kernel(const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
                 void* __restrict__ Cout, const float* __restrict__ bias,
                 int M, int N, int K, float* __restrict__ ws, int ksplit) {
  __shared__ bf16_t lds[GLDS_ELEMS];
  const int m0 = blockIdx.x * GBM;
  const int n0 = blockIdx.y * GBN;
  const int t = threadIdx.x;
  const int wave = t >> 6, wr = wave >> 1, wc = wave & 1;

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0, 0, 0, 0};

  // reduction length per mode (MODE1 reduces N, MODE2 reduces M): the
  // round-1 kernel looped cdiv(K) steps for every mode — on the head
  // backward that was 64 K-steps with only 1-2 carrying data (rest masked
  // to zero), most of the measured 40-166 us
  const int red = (MODE == 0) ? K : (MODE == 1 ? N : M);
  const int nk = cdiv_i(red, GBK);
  const int steps = (nk + ksplit - 1) / ksplit;
  const int kt0 = blockIdx.z * steps;
  const int kt1 = min(kt0 + steps, nk);
  // staging: thread t loads 8 elems of At and 8 of Bt: row = t&63, kk = t>>6+4i
  const int srow = t & 63, sk0 = t >> 6;

  for (int kt = kt0; kt < kt1; ++kt) {
    const int k0 = kt * GBK;
    bf16_t* At = &lds[GAS_OFF(kt & 1)];
    bf16_t* Bt = &lds[GBS_OFF(kt & 1)];
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      int kk = sk0 + i * 4;
      int k = k0 + kk;
      bf16_t av = 0, bv = 0;
      if (MODE == 0) {          // A[M,K] row-major, B = W[N,K] row-major
        if (m0 + srow < M && k < K) av = A[(long)(m0 + srow) * K + k];
        if (n0 + srow < N && k < K) bv = B[(long)(n0 + srow) * K + k];
      } else if (MODE == 1) {   // A = dY[M,N], B = W[N,K]: out dX[M,K]
        // GEMM dims: rows M, cols K(out), reduce N -> "k" here is the N axis
        if (m0 + srow < M && k < N) av = A[(long)(m0 + srow) * N + k];
        if (n0 + srow < K && k < N) bv = B[(long)k * K + n0 + srow];
      } else {                  // A = dY[M,N]^T -> rows N; B = X[M,K]; reduce M
        if (m0 + srow < N && k < M) av = A[(long)k * N + m0 + srow];
        if (n0 + srow < K && k < M) bv = B[(long)k * K + n0 + srow];
      }
      At[srow * GLP + kk] = av;
      Bt[srow * GLP + kk] = bv;
    }
    __syncthreads();
    {
      FragIdx fi = frag_idx();
      const int kb = fi.quad * 8;
#pragma unroll
      for (int mr = 0; mr < 2; ++mr) {
        bf16x8 a = *(const bf16x8*)&At[(wr * 32 + mr * 16 + fi.half) * GLP + kb];
#pragma unroll
        for (int nr = 0; nr < 2; ++nr) {
          bf16x8 b = *(const bf16x8*)&Bt[(wc * 32 + nr * 16 + fi.half) * GLP
                                         + kb];
          acc[mr][nr] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b,
                                                                acc[mr][nr],
                                                                0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  const int outM = (MODE == 0) ? M : (MODE == 1 ? M : N);
  const int outN = (MODE == 0) ? N : K;
  FragIdx fi = frag_idx();
#pragma unroll
  for (int mr = 0; mr < 2; ++mr)
#pragma unroll
    for (int nr = 0; nr < 2; ++nr)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr * 32 + mr * 16 + fi.quad * 4 + r;
        int col = n0 + wc * 32 + nr * 16 + fi.half;
        if (row < outM && col < outN) {
          float v = acc[mr][nr][r];
          // bias folds into slice 0's partial under split-K (deterministic)
          if (MODE == 0 && bias != nullptr && blockIdx.z == 0)
            v += bias[col];
          if (ksplit > 1)
            ws[((long)blockIdx.z * outM + row) * outN + col] = v;
          else if (OUT_F32)
            ((float*)Cout)[(long)row * outN + col] = v;
          else
            ((bf16_t*)Cout)[(long)row * outN + col] = f2bf(v);
        }
      }
}

// db[n] = sum_m dy[m][n]  (fp32 out)
__global__ __launch_bounds__(256)
void colsum_kernel(const bf16_t* __restrict__ dy, float* __restrict__ db,
                   int M, int N) {
  int n = blockIdx.x * 256 + threadIdx.x;
  if (n >= N) return;
  float s = 0.f;
  for (int m = 0; m < M; ++m) s += bf2f(dy[(long)m * N + n]);
  db[n] = s;
}

extern "C" {

int cilfw_linear_ksplit(int rows, int cols, int red) {
  // fill ~256 CUs: tiny head tiles get K-sliced (>= 4 K-steps per slice)
  int blocks = cdiv(rows, GBM) * cdiv(cols, GBN);
  int nk = cdiv(red, GBK);
  int ks = 1;
  while (ks < 32 && blocks * ks < 256 && nk / (ks * 2) >= 4) ks *= 2;
  return ks;
}

void cilfw_linear_fwd(const void* x, const void* w, const void* bias, void* y,
                      void* ws, int M, int N, int K, int ksplit,
                      void* stream) {
  dim3 grid(cdiv(M, GBM), cdiv(N, GBN), ksplit);
  hipLaunchKernelGGL((gemm_kernel<0, false>), grid, dim3(GNT), 0,
                     (hipStream_t)stream, (const bf16_t*)x, (const bf16_t*)w,
                     y, (const float*)bias, M, N, K, (float*)ws, ksplit);
  if (ksplit > 1) {
    long len = (long)M * N;
    hipLaunchKernelGGL(reduce_slabs_bf16_kernel,
                       dim3((int)cdiv((long)len, 1024L)), dim3(256), 0,
                       (hipStream_t)stream, (float*)ws, (bf16_t*)y, ksplit,
                       len);
  }
}

void cilfw_linear_dx(const void* dy, const void* w, void* dx, void* ws,
                     int M, int N, int K, int ksplit, void* stream) {
  dim3 grid(cdiv(M, GBM), cdiv(K, GBN), ksplit);
  hipLaunchKernelGGL((gemm_kernel<1, false>), grid, dim3(GNT), 0,
                     (hipStream_t)stream, (const bf16_t*)dy, (const bf16_t*)w,
                     dx, nullptr, M, N, K, (float*)ws, ksplit);
  if (ksplit > 1) {
    long len = (long)M * K;
    hipLaunchKernelGGL(reduce_slabs_bf16_kernel,
                       dim3((int)cdiv((long)len, 1024L)), dim3(256), 0,
                       (hipStream_t)stream, (float*)ws, (bf16_t*)dx, ksplit,
                       len);
  }
}

void cilfw_linear_dw(const void* dy, const void* x, void* dw, void* db,
                     void* ws, int M, int N, int K, int ksplit,
                     void* stream) {
  dim3 grid(cdiv(N, GBM), cdiv(K, GBN), ksplit);
  hipLaunchKernelGGL((gemm_kernel<2, true>), grid, dim3(GNT), 0,
                     (hipStream_t)stream, (const bf16_t*)dy, (const bf16_t*)x,
                     dw, nullptr, M, N, K, (float*)ws, ksplit);
  if (ksplit > 1) {
    long len = (long)N * K;
    hipLaunchKernelGGL(reduce_slabs_f32_kernel,
                       dim3((int)cdiv((long)len, 1024L)), dim3(256), 0,
                       (hipStream_t)stream, (float*)ws, (float*)dw, ksplit,
                       len, 0);
  }
  if (db != nullptr)
    hipLaunchKernelGGL(colsum_kernel, dim3(cdiv(N, 256)), dim3(256), 0,
                       (hipStream_t)stream, (const bf16_t*)dy, (float*)db, M,
                       N);
}

}  // extern "C"
