// Fused rotary position embedding apply (fwd + bwd), bf16.
//
// t: [s, b, nh, d]; cos/sin tables: [s, d] fp32 with the duplicated-half
// layout (cos[k+d/2] == cos[k]) precomputed on host (guide Appendix B:
// never evaluate trig on the VALU in an HBM-bound op).  Each thread
// processes 4 (lo, hi) element pairs: out_lo = lo*c - hi*s;
// out_hi = hi*c + lo*s.  Backward is the transposed rotation.

#include "common.h"

#include <stdexcept>

#define BLOCK 256
#define PVEC 4  // pairs per thread

template <bool BWD>
__global__ void rope_kernel(const unsigned short* __restrict__ t,
                            const float* __restrict__ cs,
                            const float* __restrict__ sn,
                            unsigned short* __restrict__ out, long rows,
                            int bnh, int d) {
  // rows = s*b*nh; seq index = row / bnh
  const int half = d / 2;
  long idx = (long)blockIdx.x * BLOCK + threadIdx.x;
  const long total = rows * (half / PVEC);
  const long stride = (long)gridDim.x * BLOCK;
  for (; idx < total; idx += stride) {
    const long row = idx / (half / PVEC);
    const int k = (int)(idx % (half / PVEC)) * PVEC;
    const long srow = row / bnh;
    const unsigned short* tr = t + row * d;
    unsigned short* orow = out + row * d;
    short4v lo = *(const short4v*)(tr + k);
    short4v hi = *(const short4v*)(tr + half + k);
    float4v c = *(const float4v*)(cs + srow * d + k);
    float4v s = *(const float4v*)(sn + srow * d + k);
    short4v olo, ohi;
#pragma unroll
    for (int j = 0; j < PVEC; ++j) {
      float a = bf2f((unsigned short)lo[j]);
      float b = bf2f((unsigned short)hi[j]);
      if (!BWD) {
        olo[j] = (short)f2bf(a * c[j] - b * s[j]);
        ohi[j] = (short)f2bf(b * c[j] + a * s[j]);
      } else {
        olo[j] = (short)f2bf(a * c[j] + b * s[j]);
        ohi[j] = (short)f2bf(b * c[j] - a * s[j]);
      }
    }
    *(short4v*)(orow + k) = olo;
    *(short4v*)(orow + half + k) = ohi;
  }
}

void launch_rope(const void* t, const float* cs, const float* sn, void* out,
                 long rows, int bnh, int d, bool bwd, hipStream_t stream) {
  if ((d / 2) % PVEC != 0) throw std::runtime_error("rot_dim/2 must be divisible by 4");
  long work = rows * (d / 2 / PVEC);
  long blocks = (work + BLOCK - 1) / BLOCK;
  int grid = (int)(blocks < 2048 ? (blocks < 1 ? 1 : blocks) : 2048);
  if (bwd)
    hipLaunchKernelGGL(rope_kernel<true>, dim3(grid), dim3(BLOCK), 0, stream,
                       (const unsigned short*)t, cs, sn, (unsigned short*)out,
                       rows, bnh, d);
  else
    hipLaunchKernelGGL(rope_kernel<false>, dim3(grid), dim3(BLOCK), 0, stream,
                       (const unsigned short*)t, cs, sn, (unsigned short*)out,
                       rows, bnh, d);
  HIP_CHECK_LAUNCH();
}
