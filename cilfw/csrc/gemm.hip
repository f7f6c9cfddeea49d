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

// MODE: 0 = NT (fwd), 1 = NN (dx), 2 = TN (dw)
template <int MODE, bool OUT_F32>
__global__ __launch_bounds__(GNT)
void gemm_kernel(const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
                 void* __restrict__ Cout, const float* __restrict__ bias,
                 int M, int N, int K) {
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

  const int nk = cdiv_i(K, GBK);
  // staging: thread t loads 8 elems of At and 8 of Bt: row = t&63, kk = t>>6+4i
  const int srow = t & 63, sk0 = t >> 6;

  for (int kt = 0; kt < nk; ++kt) {
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
          if (MODE == 0 && bias != nullptr) v += bias[col];
          if (OUT_F32)
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

void cilfw_linear_fwd(const void* x, const void* w, const void* bias, void* y,
                      int M, int N, int K, void* stream) {
  dim3 grid(cdiv(M, GBM), cdiv(N, GBN));
  hipLaunchKernelGGL((gemm_kernel<0, false>), grid, dim3(GNT), 0,
                     (hipStream_t)stream, (const bf16_t*)x, (const bf16_t*)w,
                     y, (const float*)bias, M, N, K);
}

void cilfw_linear_dx(const void* dy, const void* w, void* dx, int M, int N,
                     int K, void* stream) {
  dim3 grid(cdiv(M, GBM), cdiv(K, GBN));
  hipLaunchKernelGGL((gemm_kernel<1, false>), grid, dim3(GNT), 0,
                     (hipStream_t)stream, (const bf16_t*)dy, (const bf16_t*)w,
                     dx, nullptr, M, N, K);
}

void cilfw_linear_dw(const void* dy, const void* x, void* dw, void* db,
                     int M, int N, int K, void* stream) {
  dim3 grid(cdiv(N, GBM), cdiv(K, GBN));
  hipLaunchKernelGGL((gemm_kernel<2, true>), grid, dim3(GNT), 0,
                     (hipStream_t)stream, (const bf16_t*)dy, (const bf16_t*)x,
                     dw, nullptr, M, N, K);
  if (db != nullptr)
    hipLaunchKernelGGL(colsum_kernel, dim3(cdiv(N, 256)), dim3(256), 0,
                       (hipStream_t)stream, (const bf16_t*)dy, (float*)db, M,
                       N);
}

}  // extern "C"
