// Empirical probe of ds_read_b64_tr_b16 addressing on gfx950:
// LDS filled with element index; dump what each lane receives under three
// per-lane address conventions.
#include <hip/hip_runtime.h>
#include <stdio.h>
typedef __attribute__((ext_vector_type(4))) short s4;

__global__ void trprobe(short* out) {
  __shared__ short lds[1024];
  for (int i = threadIdx.x; i < 1024; i += 64) lds[i] = (short)i;
  __syncthreads();
  int l = threadIdx.x;
  unsigned base = (unsigned)(uintptr_t)&lds[0];
  unsigned a1 = base + 2u * (l & 15);        // element-offset convention
  unsigned a2 = base + 8u * (l & 15);        // 8-B slot convention
  unsigned a3 = base;                        // uniform
  s4 v1, v2, v3;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=&v"(v1) : "v"(a1));
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=&v"(v2) : "v"(a2));
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=&v"(v3) : "v"(a3));
  for (int j = 0; j < 4; ++j) {
    out[l * 12 + j] = v1[j];
    out[l * 12 + 4 + j] = v2[j];
    out[l * 12 + 8 + j] = v3[j];
  }
}

int main() {
  short* d;
  hipMalloc(&d, 64 * 12 * sizeof(short));
  hipLaunchKernelGGL(trprobe, dim3(1), dim3(64), 0, 0, d);
  short h[64 * 12];
  hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  for (const char* name : {"elem-offset", "slot8", "uniform"}) (void)name;
  const char* names[3] = {"elem-offset(+2*(l&15))", "slot8(+8*(l&15))", "uniform"};
  for (int conv = 0; conv < 3; ++conv) {
    printf("== %s ==\n", names[conv]);
    for (int l = 0; l < 64; ++l) {
      printf("l%02d:", l);
      for (int j = 0; j < 4; ++j) printf(" %4d", h[l * 12 + conv * 4 + j]);
      printf(l % 4 == 3 ? "\n" : "  |");
    }
  }
  return 0;
}
