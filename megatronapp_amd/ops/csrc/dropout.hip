// Fused bias + dropout + residual-add with on-device philox RNG
// (reference fused_bias_dropout.py's train path; the eager composition
// is 3 HBM-bound kernels + an RNG-state fork).
//
//   out  = residual + mask * (x + bias) / (1 - p)
//   mask ~ Bernoulli(1 - p) from rocrand philox4x32_10(seed, idx)
//
// The byte mask is stored so the backward is one pass:
//   dx = dy * mask / (1 - p)
// (bias grad goes through the caller's colsum path, residual grad = dy.)

#include "common.h"

#include <rocrand/rocrand_kernel.h>
#include <stdexcept>

typedef __attribute__((ext_vector_type(8))) short short8v_d;

#define DO_BLOCK 256
#define DO_VEC 4  // philox yields 4 uniforms per call

__global__ __launch_bounds__(DO_BLOCK) void bias_dropout_add_fwd_kernel(
    const unsigned short* __restrict__ x,
    const unsigned short* __restrict__ bias,
    const unsigned short* __restrict__ residual,
    unsigned short* __restrict__ out, unsigned char* __restrict__ mask,
    long n, int F, float p, unsigned long long seed) {
  const float keep_inv = 1.0f / (1.0f - p);
  long i = ((long)blockIdx.x * DO_BLOCK + threadIdx.x) * DO_VEC;
  const long stride = (long)gridDim.x * DO_BLOCK * DO_VEC;
  for (; i < n; i += stride) {
    rocrand_state_philox4x32_10 st;
    rocrand_init(seed, (unsigned long long)(i / DO_VEC), 0, &st);
    float4 r = rocrand_uniform4(&st);
    const float rs[4] = {r.x, r.y, r.z, r.w};
#pragma unroll
    for (int j = 0; j < DO_VEC; ++j) {
      const long k = i + j;
      if (k >= n) break;
      const unsigned char keep = rs[j] > p ? 1 : 0;
      float v = bf2f(x[k]);
      if (bias != nullptr) v += bf2f(bias[(int)(k % F)]);
      v = keep ? v * keep_inv : 0.0f;
      out[k] = f2bf(v + bf2f(residual[k]));
      mask[k] = keep;
    }
  }
}

__global__ __launch_bounds__(DO_BLOCK) void dropout_bwd_kernel(
    const unsigned short* __restrict__ dy,
    const unsigned char* __restrict__ mask, unsigned short* __restrict__ dx,
    long n, float p) {
  const float keep_inv = 1.0f / (1.0f - p);
  long i = (long)blockIdx.x * DO_BLOCK + threadIdx.x;
  const long stride = (long)gridDim.x * DO_BLOCK;
  for (; i < n; i += stride)
    dx[i] = mask[i] ? f2bf(bf2f(dy[i]) * keep_inv) : (unsigned short)0;
}

static int do_grid(long work) {
  long blocks = (work + DO_BLOCK - 1) / DO_BLOCK;
  return (int)(blocks < 4096 ? (blocks < 1 ? 1 : blocks) : 4096);
}

void launch_bias_dropout_add_fwd(const void* x, const void* bias,
                                 const void* residual, void* out,
                                 unsigned char* mask, long n, int F, float p,
                                 unsigned long long seed,
                                 hipStream_t stream) {
  if (p <= 0.0f || p >= 1.0f)
    throw std::runtime_error("bias_dropout_add: p must be in (0, 1)");
  hipLaunchKernelGGL(bias_dropout_add_fwd_kernel,
                     dim3(do_grid(n / DO_VEC)), dim3(DO_BLOCK), 0, stream,
                     (const unsigned short*)x, (const unsigned short*)bias,
                     (const unsigned short*)residual, (unsigned short*)out,
                     mask, n, F, p, seed);
  HIP_CHECK_LAUNCH();
}

void launch_dropout_bwd(const void* dy, const unsigned char* mask, void* dx,
                        long n, float p, hipStream_t stream) {
  hipLaunchKernelGGL(dropout_bwd_kernel, dim3(do_grid(n)), dim3(DO_BLOCK),
                     0, stream, (const unsigned short*)dy, mask,
                     (unsigned short*)dx, n, p);
  HIP_CHECK_LAUNCH();
}
