// Fused vocab-parallel cross entropy (core/tensor_parallel/cross_entropy.py
// keeps the TP all-reduce structure; these kernels replace its eager torch
// passes, which materialize an fp32 softmax the size of the logits).
// Layout: one block per token row, short8 (bf16x8) loads over the vocab
// shard.  fp32 row statistics; the bf16 logits are the only big tensor
// read, and backward writes dlogits straight in bf16.
#include "common.h"

#define CE_BLOCK 256

__global__ void ce_rowmax_kernel(const unsigned short* __restrict__ logits,
                                 float* __restrict__ rowmax, long rows,
                                 int v) {
  __shared__ float lds[CE_BLOCK / WAVE];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* lp = logits + row * v;
    float m = -INFINITY;
    for (int base = threadIdx.x * 8; base < v; base += CE_BLOCK * 8) {
      short8v x = *(const short8v*)(lp + base);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        m = fmaxf(m, bf2f((unsigned short)x[j]));
    }
    m = block_reduce_max<CE_BLOCK>(m, lds);
    if (threadIdx.x == 0) rowmax[row] = m;
    __syncthreads();
  }
}

// sumexp(x - gmax) per row + the shifted target logit (targets already
// shard-local; < 0 means the target lives on another TP rank)
__global__ void ce_fwd_kernel(const unsigned short* __restrict__ logits,
                              const float* __restrict__ rowmax,
                              const int* __restrict__ target,
                              float* __restrict__ sumexp,
                              float* __restrict__ predicted, long rows,
                              int v) {
  __shared__ float lds[CE_BLOCK / WAVE];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* lp = logits + row * v;
    const float m = rowmax[row];
    const int tgt = target[row];
    float sum = 0.f;
    for (int base = threadIdx.x * 8; base < v; base += CE_BLOCK * 8) {
      short8v x = *(const short8v*)(lp + base);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float xf = bf2f((unsigned short)x[j]) - m;
        sum += __expf(xf);
        if (base + j == tgt) predicted[row] = xf;
      }
    }
    sum = block_reduce_sum<CE_BLOCK>(sum, lds);
    if (threadIdx.x == 0) {
      sumexp[row] = sum;
      if (tgt < 0) predicted[row] = 0.f;
    }
    __syncthreads();
  }
}

// dlogits = (softmax - onehot) * grad_row, written bf16 in place of a
// separate fp32 softmax tensor
__global__ void ce_bwd_kernel(const unsigned short* __restrict__ logits,
                              const float* __restrict__ rowmax,
                              const float* __restrict__ sumexp,
                              const int* __restrict__ target,
                              const float* __restrict__ grad_row,
                              unsigned short* __restrict__ dlogits, long rows,
                              int v) {
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* lp = logits + row * v;
    unsigned short* dp = dlogits + row * v;
    const float m = rowmax[row];
    const float inv = 1.f / sumexp[row];
    const float g = grad_row[row];
    const int tgt = target[row];
    for (int base = threadIdx.x * 8; base < v; base += CE_BLOCK * 8) {
      short8v x = *(const short8v*)(lp + base);
      short8v o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float p = __expf(bf2f((unsigned short)x[j]) - m) * inv;
        if (base + j == tgt) p -= 1.f;
        o[j] = (short)f2bf(p * g);
      }
      *(short8v*)(dp + base) = o;
    }
  }
}

static int ce_grid(long rows) {
  return (int)(rows < 4096 ? (rows < 1 ? 1 : rows) : 4096);
}

void launch_ce_rowmax(const void* logits, float* rowmax, long rows, int v,
                      hipStream_t s) {
  hipLaunchKernelGGL(ce_rowmax_kernel, dim3(ce_grid(rows)), dim3(CE_BLOCK), 0,
                     s, (const unsigned short*)logits, rowmax, rows, v);
  HIP_CHECK_LAUNCH();
}

void launch_ce_fwd(const void* logits, const float* rowmax, const int* target,
                   float* sumexp, float* predicted, long rows, int v,
                   hipStream_t s) {
  hipLaunchKernelGGL(ce_fwd_kernel, dim3(ce_grid(rows)), dim3(CE_BLOCK), 0, s,
                     (const unsigned short*)logits, rowmax, target, sumexp,
                     predicted, rows, v);
  HIP_CHECK_LAUNCH();
}

void launch_ce_bwd(const void* logits, const float* rowmax,
                   const float* sumexp, const int* target,
                   const float* grad_row, void* dlogits, long rows, int v,
                   hipStream_t s) {
  hipLaunchKernelGGL(ce_bwd_kernel, dim3(ce_grid(rows)), dim3(CE_BLOCK), 0, s,
                     (const unsigned short*)logits, rowmax, sumexp, target,
                     grad_row, (unsigned short*)dlogits, rows, v);
  HIP_CHECK_LAUNCH();
}
