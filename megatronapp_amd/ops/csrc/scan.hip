// Selective state-space scan, sequential form (core/ssm/selective_scan.py
// dispatches here for no-grad decode; the chunk-parallel torch form covers
// training).  One lane per (batch, channel) row: the N states live in
// registers and the t-loop streams x/dt coalesced across adjacent channel
// lanes.  B/C rows are shared by every channel of a batch -> L2-served.
#include "common.h"

template <int N>
__global__ void selective_scan_fwd_kernel(
    const unsigned short* __restrict__ x,   // [b, l, d] bf16
    const unsigned short* __restrict__ dt,  // [b, l, d] bf16 (softplus'd)
    const float* __restrict__ A,            // [d, N]
    const unsigned short* __restrict__ B,   // [b, l, N] bf16
    const unsigned short* __restrict__ C,   // [b, l, N] bf16
    const float* __restrict__ Dp,           // [d]
    float* __restrict__ h,                  // [b, d, N] inout fp32
    unsigned short* __restrict__ y,         // [b, l, d] bf16
    int b, int l, int d) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= (long)b * d) return;
  const int bi = (int)(i / d);
  const int di = (int)(i % d);

  float hs[N];
  float Ar[N];
#pragma unroll
  for (int j = 0; j < N; ++j) {
    hs[j] = h[((long)bi * d + di) * N + j];
    Ar[j] = A[(long)di * N + j];
  }
  const float Dv = Dp[di];
  const unsigned short* xp = x + (long)bi * l * d + di;
  const unsigned short* dtp = dt + (long)bi * l * d + di;
  const unsigned short* Bp = B + (long)bi * l * N;
  const unsigned short* Cp = C + (long)bi * l * N;
  unsigned short* yp = y + (long)bi * l * d + di;

  for (int t = 0; t < l; ++t) {
    const float xv = bf2f(xp[(long)t * d]);
    const float dtv = bf2f(dtp[(long)t * d]);
    float acc = 0.f;
#pragma unroll
    for (int j = 0; j < N; ++j) {
      const float Bv = bf2f(Bp[(long)t * N + j]);
      const float Cv = bf2f(Cp[(long)t * N + j]);
      hs[j] = __expf(dtv * Ar[j]) * hs[j] + dtv * Bv * xv;
      acc += Cv * hs[j];
    }
    yp[(long)t * d] = f2bf(acc + Dv * xv);
  }
#pragma unroll
  for (int j = 0; j < N; ++j) h[((long)bi * d + di) * N + j] = hs[j];
}

void launch_selective_scan_fwd(const void* x, const void* dt, const float* A,
                               const void* B, const void* C, const float* D,
                               float* h, void* y, int b, int l, int d, int n,
                               hipStream_t s) {
  const long rows = (long)b * d;
  const int block = 256;
  const int grid = (int)((rows + block - 1) / block);
#define SCAN_CASE(N)                                                          \
  case N:                                                                     \
    hipLaunchKernelGGL((selective_scan_fwd_kernel<N>), dim3(grid),            \
                       dim3(block), 0, s, (const unsigned short*)x,           \
                       (const unsigned short*)dt, A,                          \
                       (const unsigned short*)B, (const unsigned short*)C,    \
                       D, h, (unsigned short*)y, b, l, d);                    \
    break;
  switch (n) {
    SCAN_CASE(8)
    SCAN_CASE(16)
    SCAN_CASE(32)
    default:
      throw std::runtime_error("selective_scan: n must be 8, 16 or 32");
  }
#undef SCAN_CASE
  HIP_CHECK_LAUNCH();
}
