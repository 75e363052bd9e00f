// Empirical probe for gfx950 LDS transpose-read + permlane semantics.
// Prints, for each lane, which LDS element indices ds_read_b64_tr_b16
// delivers, so the attention kernels can build V images around the real
// hardware mapping instead of guessed docs.
//
// Build: hipcc --offload-arch=gfx950 -o probe_tr16 tools/probe_tr16.hip
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(4))) short s4;

__global__ void probe_tr(int* out, int addr_mode) {
  __shared__ unsigned short lds[4096];
  const int tid = threadIdx.x;  // one wave: 64
  for (int i = tid; i < 4096; i += 64) lds[i] = (unsigned short)i;
  __syncthreads();
  // per-lane element address (8B-aligned = 4-element aligned)
  int elem;
  switch (addr_mode) {
    case 0: elem = tid * 4; break;              // linear: lane l -> elems 4l..4l+3
    case 1: elem = (tid % 16) * 4 + (tid / 16) * 64; break;  // 16-lane groups repeat
    case 2: elem = (tid % 16) * 8; break;       // stride 8 within group
    default: elem = tid * 4;
  }
  const unsigned short* p = &lds[elem];
  s4 v;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(v)
               : "v"((unsigned)(unsigned long long)(uintptr_t)p));
#pragma unroll
  for (int j = 0; j < 4; ++j) out[tid * 4 + j] = (int)(unsigned short)v[j];
}

__global__ void probe_permlane(int* out) {
  const int tid = threadIdx.x;
  unsigned a = 1000 + tid;   // value identifies source lane
  unsigned b = 2000 + tid;
  auto r = __builtin_amdgcn_permlane32_swap(a, b, false, false);
  out[tid] = (int)r[0];
  out[64 + tid] = (int)r[1];
}

__global__ void probe_cvtpk(unsigned* out) {
  // confirm v_cvt_pk_bf16_f32 packs (lo, hi) -> bf16(lo) | bf16(hi)<<16
  float lo = 1.5f, hi = -2.25f;
  unsigned p;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(p) : "v"(lo), "v"(hi));
  out[threadIdx.x] = p;
}

int main() {
  int* d;
  hipMalloc(&d, 4096);
  int h[256];
  for (int mode = 0; mode < 3; ++mode) {
    hipLaunchKernelGGL(probe_tr, dim3(1), dim3(64), 0, 0, d, mode);
    hipMemcpy(h, d, 256 * sizeof(int), hipMemcpyDeviceToHost);
    printf("== tr16 mode %d (lane: j0 j1 j2 j3) ==\n", mode);
    for (int l = 0; l < 64; ++l) {
      printf("%2d: %4d %4d %4d %4d%s", l, h[4 * l], h[4 * l + 1],
             h[4 * l + 2], h[4 * l + 3], (l % 4 == 3) ? "\n" : "   ");
    }
  }
  hipLaunchKernelGGL(probe_permlane, dim3(1), dim3(64), 0, 0, d);
  hipMemcpy(h, d, 128 * sizeof(int), hipMemcpyDeviceToHost);
  printf("== permlane32_swap r0 (lane: val) ==\n");
  for (int l = 0; l < 64; ++l)
    printf("%2d:%4d%s", l, h[l], (l % 8 == 7) ? "\n" : " ");
  printf("== permlane32_swap r1 ==\n");
  for (int l = 0; l < 64; ++l)
    printf("%2d:%4d%s", l, h[64 + l], (l % 8 == 7) ? "\n" : " ");
  unsigned* du;
  hipMalloc(&du, 256);
  hipLaunchKernelGGL(probe_cvtpk, dim3(1), dim3(64), 0, 0, du);
  unsigned hu[64];
  hipMemcpy(hu, du, 64 * 4, hipMemcpyDeviceToHost);
  printf("cvt_pk_bf16_f32(1.5, -2.25) = 0x%08x (expect 0x3fc0 | 0xc010<<16)\n",
         hu[0]);
  return 0;
}
