// conv fwd v2 experiment — all-glds staging for the implicit-GEMM conv.
//
// Structure (guide T3 "minimum 2-phase" + glds rows of the staging table):
//   A tile 128(M)x64(K): global_load_lds with per-lane GATHER source
//     addresses (zero page for OOB), XOR source swizzle so fragment
//     ds_read_b128 are conflict-free over the linear LDS image.
//   B tile 64(K)x64(N): global_load_lds into [ngrp][k][16] images consumed
//     by ds_read_b64_tr_b16 pairs (w is k-major; glds cannot transpose).
//   2-phase loop: STAGE(next) -> ds_read fragments -> lgkmcnt(0) ->
//     setprio(1) MFMA setprio(0) -> vmcnt(0) -> raw barrier.
// Compares numerics vs a CPU oracle and times the ResNet shapes.
#include <hip/hip_runtime.h>
#include <stdio.h>
#include <stdlib.h>
#include <vector>
#include <cmath>
#include <cstring>

typedef unsigned short bf16_t;
typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(4))) short s4;

static inline float bf2f_h(bf16_t v) { union { unsigned u; float f; } x; x.u = ((unsigned)v) << 16; return x.f; }
static inline bf16_t f2bf_h(float f) { union { unsigned u; float f; } x; x.f = f; unsigned r = (x.u + 0x7fff + ((x.u >> 16) & 1)) >> 16; return (bf16_t)r; }
__device__ __forceinline__ float bf2f(bf16_t v) { union { unsigned u; float f; } x; x.u = ((unsigned)v) << 16; return x.f; }
__device__ __forceinline__ bf16_t f2bf(float f) { union { unsigned u; float f; } x; x.f = f; unsigned r = (x.u + 0x7fffu + ((x.u >> 16) & 1u)) >> 16; return (bf16_t)r; }

#define BM 128
#define BN 64
#define BK 64
#define NT 256
#ifndef NBUFS
#define NBUFS 2          // 2 = vmcnt(0)+barrier; 3 = counted vmcnt(6), tile in flight
#endif

struct Geom { int N, H, W, C, K, R, S, stride, pad, Ho, Wo; };

// LDS layout (elements): A [128][64] linear (glds dest), byte-swizzled reads;
// B [4 ngrp][64 k][16 n].  Double buffered.
#define A_ELEMS (BM * BK)
#define B_ELEMS (4 * BK * 16)
#define BUF_ELEMS (A_ELEMS + B_ELEMS)

__global__ __launch_bounds__(NT)
void convfwd_v2(const bf16_t* __restrict__ x, const bf16_t* __restrict__ w,
                bf16_t* __restrict__ y, const bf16_t* __restrict__ zpage,
                Geom g, int M, int CRS, int nk) {
  __shared__ bf16_t lds[NBUFS * BUF_ELEMS];
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int t = threadIdx.x;
  const int wv = t >> 6, lane = t & 63;
  const int wr = wv >> 1, wc = wv & 1;

  // ---- A staging geometry: 16 glds covering 8 rows each; this wave does
  // instrs wv*4..wv*4+3 -> rows [wv*32 + i*8, +8). Lane covers row
  // wv*32 + i*8 + l/8, source k-chunk = ((l%8) ^ (row&7)) (XOR source
  // swizzle; read side XORs the same way).
  int arow[4], achunk[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    arow[i] = wv * 32 + i * 8 + (lane >> 3);
    achunk[i] = (lane & 7) ^ (arow[i] & 7);
  }
  // per-instr incremental (r,s,c) decomposition of k = kt*64 + achunk*8
  int ar[4], as_[4], ac[4];
  bool mok[4];
  int an[4], ahb[4], awb[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int m = m0 + arow[i];
    mok[i] = m < M;
    int mc = m < M ? m : M - 1;
    int n = mc / (g.Ho * g.Wo);
    int rem = mc - n * (g.Ho * g.Wo);
    an[i] = n;
    ahb[i] = (rem / g.Wo) * g.stride - g.pad;
    awb[i] = (rem - (rem / g.Wo) * g.Wo) * g.stride - g.pad;
    int k = achunk[i] * 8;
    int rs = k / g.C;
    ac[i] = k - rs * g.C;
    ar[i] = rs / g.S;
    as_[i] = rs - ar[i] * g.S;
  }
  // ---- B staging: 8 glds (2 per wave). Wave wv owns ngrp wv; instr j
  // covers k rows [j*32, j*32+32), lane -> k = j*32 + l/2, nsub = (l%2)*8.
  int bk[2];
#pragma unroll
  for (int j = 0; j < 2; ++j) bk[j] = j * 32 + (lane >> 1);
  const int bns = (lane & 1) * 8;

  f32x4 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i) { acc[i][0] = f32x4{0,0,0,0}; acc[i][1] = f32x4{0,0,0,0}; }

  auto stage = [&](int buf, int kt) {
    bf16_t* base = &lds[buf * BUF_ELEMS];
    // A: 4 glds
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int hi = ahb[i] + ar[i], wi = awb[i] + as_[i];
      int k = kt * BK + achunk[i] * 8;
      bool ok = mok[i] & (k < CRS) & ((unsigned)hi < (unsigned)g.H)
                & ((unsigned)wi < (unsigned)g.W);
      const bf16_t* src = ok
          ? &x[(((long)an[i] * g.H + hi) * g.W + wi) * g.C + ac[i]]
          : zpage;
      // dest: wave-uniform base for this instr; lane lands at +lane*16B
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)
              &base[(wv * 4 + i) * 8 * BK],
          16, 0, 0);
      // advance (r,s,c) by BK
      int c = ac[i] + BK;
      int r = ar[i], s = as_[i];
      while (c >= g.C) { c -= g.C; if (++s == g.S) { s = 0; ++r; } }
      ac[i] = c; ar[i] = r; as_[i] = s;
    }
    // B: 2 glds into [ngrp=wv][k][16]
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int k = kt * BK + bk[j];
      int kc = k < CRS ? k : CRS - 1;
      bool ok = (k < CRS) & (n0 + wv * 16 + bns + 8 <= g.K);
      const bf16_t* src = ok ? &w[(long)kc * g.K + n0 + wv * 16 + bns]
                             : zpage;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)
              &base[A_ELEMS + wv * (BK * 16) + j * 32 * 16],
          16, 0, 0);
    }
  };

  const int fh = lane & 15, fq = lane >> 4;  // fragment half/quad
  const unsigned lds0 = (unsigned)(uintptr_t)&lds[0];
  // B tr-read lane base (see bwd-weight kernel): k-row fq*8 + (l>>2)&3,
  // byte slot (l&3)*8 within the 32 B row of the [k][16] image
  const unsigned btr_e = (unsigned)((fq * 8 + ((lane >> 2) & 3)) * 16
                                    + (lane & 3) * 4);

  stage(0, 0);
#if NBUFS == 3
  if (nk > 1) {
    stage(1, 1);
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
#else
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
#endif
  __builtin_amdgcn_s_barrier();
  for (int kt = 0; kt < nk; ++kt) {
    int cur = kt % NBUFS;
#if NBUFS == 3
    if (kt + 2 < nk) stage((kt + 2) % NBUFS, kt + 2);
#else
    if (kt + 1 < nk) stage((kt + 1) % NBUFS, kt + 1);
#endif
    const bf16_t* As = &lds[cur * BUF_ELEMS];
    (void)0;
    const unsigned bbase = lds0
        + 2u * (cur * BUF_ELEMS + A_ELEMS);
    // B fragments: col = wc*32 + nr*16 + fh -> ngrp = wc*2 + nr
    bf16x8 bfr[2][2];  // [q(2 k-halves of 64)][nr]
#pragma unroll
    for (int q = 0; q < 2; ++q)
#pragma unroll
      for (int nr = 0; nr < 2; ++nr) {
        unsigned a = bbase + 2u * ((wc * 2 + nr) * (BK * 16))
                     + 2u * (btr_e + q * 32 * 16);
        s4 lo, hi;
        asm volatile("ds_read_b64_tr_b16 %0, %2\n\t"
                     "ds_read_b64_tr_b16 %1, %2 offset:128\n\t"
                     : "=&v"(lo), "=&v"(hi) : "v"(a));
        bfr[q][nr] = __builtin_shufflevector(lo, hi, 0,1,2,3,4,5,6,7);
      }
#pragma unroll
    for (int q = 0; q < 2; ++q) {
      const int kb = fq * 8 + q * 32;
#pragma unroll
      for (int mr = 0; mr < 4; ++mr) {
        int row = wr * 64 + mr * 16 + fh;
        int kcol = kb ^ ((row & 7) << 3);
        bf16x8 afr = *(const bf16x8*)&As[row * BK + kcol];
        if (q == 0 && mr == 0) {
          asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
          __builtin_amdgcn_sched_barrier(0);
          __builtin_amdgcn_s_setprio(1);
        }
        acc[mr][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afr, bfr[q][0], acc[mr][0], 0, 0, 0);
        acc[mr][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afr, bfr[q][1], acc[mr][1], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
#if NBUFS == 3
    // leave the newest tile's 6 glds in flight across the barrier; at the
    // tail (nothing newer staged) drain so the LAST tile is ready
    if (kt + 2 < nk) asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
#else
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
#endif
    __builtin_amdgcn_s_barrier();
  }

#pragma unroll
  for (int mr = 0; mr < 4; ++mr)
#pragma unroll
    for (int nr = 0; nr < 2; ++nr)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr * 64 + mr * 16 + fq * 4 + r;
        int col = n0 + wc * 32 + nr * 16 + fh;
        if (row < M && col < g.K)
          y[(long)row * g.K + col] = f2bf(acc[mr][nr][r]);
      }
}

// ------------------------- host driver -------------------------

static void cpu_conv(const std::vector<float>& x, const std::vector<float>& w,
                     std::vector<float>& y, Geom g) {
  for (int n = 0; n < g.N; ++n)
    for (int ho = 0; ho < g.Ho; ++ho)
      for (int wo = 0; wo < g.Wo; ++wo)
        for (int k = 0; k < g.K; ++k) {
          float acc = 0;
          for (int r = 0; r < g.R; ++r)
            for (int s = 0; s < g.S; ++s) {
              int hi = ho * g.stride - g.pad + r;
              int wi = wo * g.stride - g.pad + s;
              if (hi < 0 || hi >= g.H || wi < 0 || wi >= g.W) continue;
              for (int c = 0; c < g.C; ++c)
                acc += x[(((long)n * g.H + hi) * g.W + wi) * g.C + c] *
                       w[(((long)r * g.S + s) * g.C + c) * g.K + k];
            }
          y[((long)(n * g.Ho + ho) * g.Wo + wo) * g.K + k] = acc;
        }
}

static double run_shape(int N, int H, int C, int K, int R, int stride,
                        bool check) {
  Geom g{N, H, H, C, K, R, R, stride, R / 2, 0, 0};
  g.Ho = (g.H + 2 * g.pad - g.R) / g.stride + 1;
  g.Wo = g.Ho;
  int M = g.N * g.Ho * g.Wo;
  int CRS = g.C * g.R * g.S;
  int nk = (CRS + BK - 1) / BK;
  std::vector<bf16_t> hx((long)N * H * H * C), hw((long)CRS * K);
  std::vector<float> fx(hx.size()), fw(hw.size());
  srand(42);
  for (size_t i = 0; i < hx.size(); ++i) { float v = (rand() % 200 - 100) / 100.f; hx[i] = f2bf_h(v); fx[i] = bf2f_h(hx[i]); }
  for (size_t i = 0; i < hw.size(); ++i) { float v = (rand() % 200 - 100) / 500.f; hw[i] = f2bf_h(v); fw[i] = bf2f_h(hw[i]); }
  bf16_t *dx, *dw, *dy, *dz;
  hipMalloc(&dx, hx.size() * 2); hipMalloc(&dw, hw.size() * 2);
  hipMalloc(&dy, (long)M * K * 2); hipMalloc(&dz, 4096);
  hipMemset(dz, 0, 4096);
  hipMemcpy(dx, hx.data(), hx.size() * 2, hipMemcpyHostToDevice);
  hipMemcpy(dw, hw.data(), hw.size() * 2, hipMemcpyHostToDevice);
  dim3 grid((M + BM - 1) / BM, (K + BN - 1) / BN);
  hipLaunchKernelGGL(convfwd_v2, grid, dim3(NT), 0, 0, dx, dw, dy, dz, g, M,
                     CRS, nk);
  hipDeviceSynchronize();
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) { printf("LAUNCH ERR %s\n", hipGetErrorString(e)); return -1; }
  if (check) {
    std::vector<bf16_t> hy((long)M * K);
    hipMemcpy(hy.data(), dy, hy.size() * 2, hipMemcpyDeviceToHost);
    std::vector<float> ref((long)M * K);
    cpu_conv(fx, fw, ref, g);
    double maxd = 0; long bad = 0;
    for (long i = 0; i < (long)hy.size(); ++i) {
      double d = fabs(bf2f_h(hy[i]) - ref[i]);
      double tol = 0.02 + 0.03 * fabs(ref[i]);
      if (d > tol) { if (bad < 5) printf("  mismatch i=%ld got=%f want=%f\n", i, bf2f_h(hy[i]), ref[i]); ++bad; }
      if (d > maxd) maxd = d;
    }
    printf("  check: maxdiff=%.4f bad=%ld/%ld -> %s\n", maxd, bad,
           (long)hy.size(), bad == 0 ? "OK" : "FAIL");
  }
  // timing
  hipEvent_t t0, t1; hipEventCreate(&t0); hipEventCreate(&t1);
  for (int i = 0; i < 10; ++i)
    hipLaunchKernelGGL(convfwd_v2, grid, dim3(NT), 0, 0, dx, dw, dy, dz, g, M, CRS, nk);
  hipDeviceSynchronize();
  hipEventRecord(t0);
  for (int i = 0; i < 50; ++i)
    hipLaunchKernelGGL(convfwd_v2, grid, dim3(NT), 0, 0, dx, dw, dy, dz, g, M, CRS, nk);
  hipEventRecord(t1);
  hipDeviceSynchronize();
  float ms; hipEventElapsedTime(&ms, t0, t1);
  double us = ms * 1000.0 / 50;
  double tf = 2.0 * M * K * CRS / (us * 1e6);
  printf("  %dx%d C%d->K%d k%ds%d: %.1f us  %.1f TF\n", H, H, C, K, R,
         stride, us, tf);
  hipFree(dx); hipFree(dw); hipFree(dy); hipFree(dz);
  return tf;
}

int main(int argc, char** argv) {
  bool quick = argc > 1 && !strcmp(argv[1], "quick");
  printf("== correctness (small) ==\n");
  run_shape(4, 16, 64, 64, 3, 1, true);
  run_shape(2, 9, 32, 64, 3, 1, true);   // ragged M
  run_shape(4, 16, 64, 128, 3, 2, true); // strided
  if (quick) return 0;
  printf("== ResNet-18 CIFAR shapes ==\n");
  run_shape(128, 32, 64, 64, 3, 1, false);    // l1
  run_shape(128, 16, 128, 128, 3, 1, false);  // l2
  run_shape(128, 8, 256, 256, 3, 1, false);   // l3
  run_shape(128, 4, 512, 512, 3, 1, false);   // l4
  return 0;
}
