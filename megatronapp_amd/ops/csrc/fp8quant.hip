// Fused per-row e4m3 quantization for the fp8 training GEMM path:
// one kernel computes row amax AND writes the quantized rows + f32
// scales (the eager torch version was 6 kernels per call with an fp32
// materialization — 24% of fp8-bench kernel time).
//
// x [M, K] bf16 (K % 8 == 0) -> q [M, K] e4m3 (uint8 storage),
// scale [M, 1] f32 = rowmax/448.  One wave per row; the row is read
// twice (second read hits L2).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8;

#include <hip/hip_fp8.h>
#include <stdexcept>

#define FP8Q_BLOCK 256

__global__ __launch_bounds__(FP8Q_BLOCK) void quant_rows_e4m3_kernel(
    const unsigned short* __restrict__ x, unsigned char* __restrict__ q,
    float* __restrict__ scale, long M, int K) {
  const long row = (long)blockIdx.x * (FP8Q_BLOCK / WAVE) +
                   threadIdx.x / WAVE;
  if (row >= M) return;
  const int lane = threadIdx.x % WAVE;
  const unsigned short* xr = x + row * K;

  // pass 1: row absmax (vectorized bf16x8 loads)
  float amax = 0.f;
  for (int i = lane * 8; i < K; i += WAVE * 8) {
    bf16x8 v = *(const bf16x8*)(xr + i);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      amax = fmaxf(amax, fabsf(bf2f((unsigned short)v[j])));
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    amax = fmaxf(amax, __shfl_xor(amax, off, WAVE));

  const float s = fmaxf(amax / 448.0f, 1e-12f);
  const float inv = 1.0f / s;
  if (lane == 0) scale[row] = s;

  // pass 2: scale + convert + store 8 bytes per iteration
  unsigned char* qr = q + row * K;
  for (int i = lane * 8; i < K; i += WAVE * 8) {
    bf16x8 v = *(const bf16x8*)(xr + i);
    uchar2 packed[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float lo = bf2f((unsigned short)v[2 * j]) * inv;
      float hi = bf2f((unsigned short)v[2 * j + 1]) * inv;
      __hip_fp8_e4m3 qlo(lo);
      __hip_fp8_e4m3 qhi(hi);
      packed[j] = make_uchar2(qlo.__x, qhi.__x);
    }
    *(uchar2*)(qr + i) = packed[0];
    *(uchar2*)(qr + i + 2) = packed[1];
    *(uchar2*)(qr + i + 4) = packed[2];
    *(uchar2*)(qr + i + 6) = packed[3];
  }
}

void launch_quant_rows_e4m3(const void* x, void* q, float* scale, long M,
                            int K, hipStream_t stream) {
  if (K % 8 != 0)
    throw std::runtime_error("quant_rows_e4m3: K must be a multiple of 8");
  const int rows_per_block = FP8Q_BLOCK / WAVE;
  const long blocks = (M + rows_per_block - 1) / rows_per_block;
  hipLaunchKernelGGL(quant_rows_e4m3_kernel, dim3((unsigned)blocks),
                     dim3(FP8Q_BLOCK), 0, stream,
                     (const unsigned short*)x, (unsigned char*)q, scale, M,
                     K);
  HIP_CHECK_LAUNCH();
}

// ---------------------------------------------------------------------------
// Transpose-quantize for the fp8 wgrad GEMM: both wgrad operands need
// their token (K) dimension innermost for the MFMA fp8 path, so the
// quantization pass does the transpose through an LDS tile for free.
//
//   x [R, C] bf16  ->  q [C, R] e4m3,  scale [1] f32 = amax/448
//
// (per-tensor just-in-time scaling: the amax is of THIS tensor, not a
// delayed-scaling history)

#define TQ_TILE 64

__global__ void amax_abs_kernel(const unsigned short* __restrict__ x, long n,
                                unsigned int* __restrict__ amax_bits) {
  __shared__ float lds[FP8Q_BLOCK / WAVE];
  float amax = 0.f;
  long i = ((long)blockIdx.x * FP8Q_BLOCK + threadIdx.x) * 8;
  const long stride = (long)gridDim.x * FP8Q_BLOCK * 8;
  for (; i + 7 < n; i += stride) {
    bf16x8 v = *(const bf16x8*)(x + i);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      amax = fmaxf(amax, fabsf(bf2f((unsigned short)v[j])));
  }
  if (blockIdx.x == 0 && threadIdx.x == 0)
    for (long k = (n / 8) * 8; k < n; ++k)
      amax = fmaxf(amax, fabsf(bf2f(x[k])));
  amax = block_reduce_max<FP8Q_BLOCK>(amax, lds);
  // positive-float bit patterns order like uints
  if (threadIdx.x == 0) atomicMax(amax_bits, __float_as_uint(amax));
}

__global__ void transpose_quant_e4m3_kernel(
    const unsigned short* __restrict__ x, unsigned char* __restrict__ q,
    const unsigned int* __restrict__ amax_bits, float* __restrict__ scale,
    long R, long C) {
  __shared__ unsigned short lds[TQ_TILE][TQ_TILE + 8];
  const float amax = __uint_as_float(*amax_bits);
  const float s = fmaxf(amax / 448.0f, 1e-12f);
  const float inv = 1.0f / s;
  if (blockIdx.x == 0 && blockIdx.y == 0 && threadIdx.x == 0) scale[0] = s;

  const long r0 = (long)blockIdx.y * TQ_TILE;
  const long c0 = (long)blockIdx.x * TQ_TILE;
  // load 64x64 bf16: 8 lanes x 8 elements per row, 32 rows per pass
  const int lx = threadIdx.x % 8;        // 8-wide column group
  const int ly = threadIdx.x / 8;        // 32 rows
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    const long r = r0 + ly + p * 32;
    const long c = c0 + lx * 8;
    if (r < R) {
      if (c + 7 < C) {
        bf16x8 v = *(const bf16x8*)(x + r * C + c);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          lds[ly + p * 32][lx * 8 + j] = (unsigned short)v[j];
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          lds[ly + p * 32][lx * 8 + j] =
              (c + j < C) ? x[r * C + c + j] : 0;
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) lds[ly + p * 32][lx * 8 + j] = 0;
    }
  }
  __syncthreads();
  // store 64x64 e4m3: 16 lanes x uchar4 per output row, 16 rows per pass
  const int ox = threadIdx.x % 16;       // 16-wide row-offset group
  const int oy = threadIdx.x / 16;       // 16 output rows
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    const long oc = c0 + oy + p * 16;    // output row == input column
    const long orr = r0 + ox * 4;        // output col == input row
    if (oc >= C) continue;
    unsigned char packed[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float v = bf2f(lds[ox * 4 + j][oy + p * 16]) * inv;
      __hip_fp8_e4m3 qv(v);
      packed[j] = qv.__x;
    }
    if (orr + 3 < R) {
      *(uchar4*)(q + oc * R + orr) = make_uchar4(packed[0], packed[1],
                                                packed[2], packed[3]);
    } else {
#pragma unroll
      for (int j = 0; j < 4; ++j)
        if (orr + j < R) q[oc * R + orr + j] = packed[j];
    }
  }
}

void launch_transpose_quant_e4m3(const void* x, void* q,
                                 unsigned int* amax_bits, float* scale,
                                 long R, long C, hipStream_t stream) {
  long blocks = (R * C / 8 + FP8Q_BLOCK - 1) / FP8Q_BLOCK;
  if (blocks > 4096) blocks = 4096;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(amax_abs_kernel, dim3((unsigned)blocks),
                     dim3(FP8Q_BLOCK), 0, stream,
                     (const unsigned short*)x, R * C, amax_bits);
  HIP_CHECK_LAUNCH();
  dim3 grid((unsigned)((C + TQ_TILE - 1) / TQ_TILE),
            (unsigned)((R + TQ_TILE - 1) / TQ_TILE));
  hipLaunchKernelGGL(transpose_quant_e4m3_kernel, grid, dim3(FP8Q_BLOCK), 0,
                     stream, (const unsigned short*)x, (unsigned char*)q,
                     amax_bits, scale, R, C);
  HIP_CHECK_LAUNCH();
}
