// Shared helpers for CDNA4 (gfx950) kernels.
// Wavefront = 64; blocks are multiples of 64; bf16 I/O is vectorized as
// short4/short8 (8-16 B/lane) per the CDNA4 HIP guide (G13).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64

typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(4))) float float4v;

__device__ __forceinline__ float bf2f(unsigned short u) {
  unsigned int x = ((unsigned int)u) << 16;
  return __builtin_bit_cast(float, x);
}

__device__ __forceinline__ unsigned short f2bf(float f) {
  // hardware RNE conversion: lowers to ONE v_cvt_pk_bf16_f32 (the manual
  // bit-math version cost 3 VALU ops and made attention VALU-bound)
  __hip_bfloat16 h = __float2bfloat16(f);
  return __builtin_bit_cast(unsigned short, h);
}

// wave-level sum over all 64 lanes
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, WAVE);
  return __shfl(v, 0, WAVE);
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, WAVE));
  return __shfl(v, 0, WAVE);
}

// block-level reduction via LDS (blockDim.x threads, <=1024)
template <int BLOCK>
__device__ __forceinline__ float block_reduce_sum(float v, float* lds) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  constexpr int NW = BLOCK / WAVE;
  float r = (threadIdx.x < NW) ? lds[threadIdx.x] : 0.f;
  if (wid == 0) {
#pragma unroll
    for (int off = NW / 2; off > 0; off >>= 1) r += __shfl_down(r, off, WAVE);
    if (lane == 0) lds[0] = r;
  }
  __syncthreads();
  float out = lds[0];
  __syncthreads();
  return out;
}

// two sums in one barrier round-trip (callers with paired row statistics
// pay 3 __syncthreads instead of 6); lds must hold 2*BLOCK/WAVE floats
template <int BLOCK>
__device__ __forceinline__ float2 block_reduce_sum2(float a, float b,
                                                    float* lds) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  a = wave_reduce_sum(a);
  b = wave_reduce_sum(b);
  constexpr int NW = BLOCK / WAVE;
  if (lane == 0) {
    lds[wid] = a;
    lds[NW + wid] = b;
  }
  __syncthreads();
  float ra = (threadIdx.x < NW) ? lds[threadIdx.x] : 0.f;
  float rb = (threadIdx.x < NW) ? lds[NW + threadIdx.x] : 0.f;
  if (wid == 0) {
#pragma unroll
    for (int off = NW / 2; off > 0; off >>= 1) {
      ra += __shfl_down(ra, off, WAVE);
      rb += __shfl_down(rb, off, WAVE);
    }
    if (lane == 0) {
      lds[0] = ra;
      lds[1] = rb;
    }
  }
  __syncthreads();
  float2 out = make_float2(lds[0], lds[1]);
  __syncthreads();
  return out;
}

template <int BLOCK>
__device__ __forceinline__ float block_reduce_max(float v, float* lds) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_reduce_max(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  constexpr int NW = BLOCK / WAVE;
  float r = (threadIdx.x < NW) ? lds[threadIdx.x] : -INFINITY;
  if (wid == 0) {
#pragma unroll
    for (int off = NW / 2; off > 0; off >>= 1)
      r = fmaxf(r, __shfl_down(r, off, WAVE));
    if (lane == 0) lds[0] = r;
  }
  __syncthreads();
  float out = lds[0];
  __syncthreads();
  return out;
}

#define HIP_CHECK_LAUNCH()                                          \
  do {                                                              \
    hipError_t e_ = hipGetLastError();                              \
    if (e_ != hipSuccess)                                           \
      throw std::runtime_error(std::string("HIP launch failed: ") + \
                               hipGetErrorString(e_));              \
  } while (0)
