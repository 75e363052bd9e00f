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

// ---- MFMA 32x32x16 bf16 layout probes -----------------------------------
typedef __attribute__((ext_vector_type(8))) short bf16x8p;
typedef __attribute__((ext_vector_type(16))) float f32x16p;

__device__ __host__ static unsigned short h_f2bf(float f) {
  unsigned u;
  __builtin_memcpy(&u, &f, 4);
  return (unsigned short)((u + 0x7fff + ((u >> 16) & 1)) >> 16);
}

// Step 1: C map via f32 MFMA 32x32x2 (A/B maps documented: A[l&31][l>>5],
// B[l>>5][l&31]).  A = one-hot column codes so C[m][n] = m*32 + n + 1.
__global__ void probe_mfma_c(float* out) {
  const int l = threadIdx.x;
  // A[i][k] = (k==0) ? i+1 : 0 ; B[k][j] = (k==0) ? 1 : 0  -> C[i][j] = i+1
  // then add a second mfma with A[i][k]=(k==0)?0:1, B=(k==1)? (j+1)/1 :0
  float a1 = ((l >> 5) == 0) ? (float)((l & 31) + 1) : 0.f;
  float b1 = ((l >> 5) == 0) ? 1.f : 0.f;
  f32x16p c;
#pragma unroll
  for (int r = 0; r < 16; ++r) c[r] = 0.f;
  c = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, c, 0, 0, 0);
  // C[m][n] = m+1 now.  Second: A=(k==1)?1000:0, B=(k==1)?(n+1):0
  float a2 = ((l >> 5) == 1) ? 1000.f : 0.f;
  float b2 = ((l >> 5) == 1) ? (float)((l & 31) + 1) : 0.f;
  c = __builtin_amdgcn_mfma_f32_32x32x2f32(a2, b2, c, 0, 0, 0);
  // C[m][n] = (m+1) + 1000*(n+1): decode -> m,n per (lane, reg)
#pragma unroll
  for (int r = 0; r < 16; ++r) out[l * 16 + r] = c[r];
}

// Step 2: B map for 32x32x16 bf16.  A = identity on k=0..15 (A[m][k] =
// (m==k)), B registers = lane-unique codes; C[k][n] = Bmat[k][n] reveals
// which (lane, j) slot the hardware reads for B[k][n].
__global__ void probe_mfma_b(float* out) {
  const int l = threadIdx.x;
  bf16x8p a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    // hypothesized A map: A[l&31][(l>>5)*8 + j]
    const int m = l & 31, kk = (l >> 5) * 8 + j;
    a[j] = (short)h_f2bf((m == kk) ? 1.f : 0.f);
    b[j] = (short)h_f2bf((float)l);        // code part 1: lane
  }
  f32x16p c;
#pragma unroll
  for (int r = 0; r < 16; ++r) c[r] = 0.f;
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  bf16x8p a2, b2;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int m = l & 31, kk = (l >> 5) * 8 + j;
    a2[j] = (short)h_f2bf((m == kk) ? 100.f : 0.f);
    b2[j] = (short)h_f2bf((float)(8 + j));  // code part 2: 100*(8+j)
  }
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a2, b2, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) out[l * 16 + r] = c[r];
}

// Step 3: A map.  B = identity (using the B map confirmed in step 2:
// B[k][n]: lane holds B[(l>>5)*8+j][l&31]), A registers = codes.
__global__ void probe_mfma_a(float* out) {
  const int l = threadIdx.x;
  bf16x8p a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int kk = (l >> 5) * 8 + j, n = l & 31;
    b[j] = (short)h_f2bf((kk == n % 16) ? 1.f : 0.f);  // B = [I16; I16]
    a[j] = (short)h_f2bf((float)l);
  }
  f32x16p c;
#pragma unroll
  for (int r = 0; r < 16; ++r) c[r] = 0.f;
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  bf16x8p a2, b2;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int kk = (l >> 5) * 8 + j, n = l & 31;
    b2[j] = (short)h_f2bf((kk == n % 16) ? 100.f : 0.f);
    a2[j] = (short)h_f2bf((float)(8 + j));
  }
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a2, b2, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) out[l * 16 + r] = c[r];
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

  // MFMA layout probes
  float* df;
  hipMalloc(&df, 64 * 16 * sizeof(float));
  float hf[1024];
  hipLaunchKernelGGL(probe_mfma_c, dim3(1), dim3(64), 0, 0, df);
  hipMemcpy(hf, df, sizeof(hf), hipMemcpyDeviceToHost);
  printf("== mfma32x32 C map: lane,reg -> (m,n) via C=m+1+1000*(n+1) ==\n");
  for (int l = 0; l < 64; l += 8) {
    for (int r = 0; r < 16; r += 4) {
      int v = (int)hf[l * 16 + r];
      printf("l%02d r%02d: m=%2d n=%2d   ", l, r, v % 1000 - 1, v / 1000 - 1);
    }
    printf("\n");
  }
  hipLaunchKernelGGL(probe_mfma_b, dim3(1), dim3(64), 0, 0, df);
  hipMemcpy(hf, df, sizeof(hf), hipMemcpyDeviceToHost);
  printf("== mfma32x32x16 B map: C[k][n]=code(lane*8+j) ==\n");
  // print where B[k][n] comes from for a few (k,n): C rows = k (0..15)
  for (int l = 0; l < 64; l += 1) {
    for (int r = 0; r < 16; ++r) {
      int code = (int)hf[l * 16 + r];
      if (r == 0 || r == 5)
        printf("lane%02d reg%02d -> code %4d (src lane %d j %d)%s", l, r,
               code, code % 100, code / 100 - 8, (r == 5) ? "\n" : "  ");
    }
  }
  hipLaunchKernelGGL(probe_mfma_a, dim3(1), dim3(64), 0, 0, df);
  hipMemcpy(hf, df, sizeof(hf), hipMemcpyDeviceToHost);
  printf("== mfma32x32x16 A map ==\n");
  for (int l = 0; l < 64; l += 1)
    for (int r = 0; r < 16; ++r)
      if (r < 2)
        printf("lane%02d reg%02d -> code %4d (src lane %d j %d)%s", l, r,
               (int)hf[l * 16 + r], (int)hf[l * 16 + r] % 100,
               (int)hf[l * 16 + r] / 100 - 8, r == 1 ? "\n" : "  ");
  return 0;
}

