// MoE token combine (the unpermute + topk-weighted sum of the a2a
// dispatcher) as one fused kernel each way (reference moe_utils.py
// fused permute/unpermute).  The eager path was index_copy + reshape-mul
// + sum: 3 full passes over [n*topk, h]; this reads each row once.
//
//  fwd: out[t] = sum_k probs[t,k] * permuted[inv_pos[t*topk + k]]
//  bwd: dpermuted[p]   = probs[t,k] * dout[t]      (p sorted; sort_idx[p]
//       dprobs[t,k]    = <permuted[p], dout[t]>     = t*topk + k)

#include "common.h"

#include <stdexcept>

typedef __attribute__((ext_vector_type(8))) short bf16x8;

#define MOE_BLOCK 256

// one wave per output token row
__global__ __launch_bounds__(MOE_BLOCK) void moe_combine_fwd_kernel(
    const unsigned short* __restrict__ permuted,
    const long* __restrict__ inv_pos, const float* __restrict__ probs,
    unsigned short* __restrict__ out, long n_tokens, int topk, int h) {
  const long t = (long)blockIdx.x * (MOE_BLOCK / WAVE) + threadIdx.x / WAVE;
  if (t >= n_tokens) return;
  const int lane = threadIdx.x % WAVE;
  for (int i = lane * 8; i < h; i += WAVE * 8) {
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int k = 0; k < topk; ++k) {
      const long p = inv_pos[t * topk + k];
      const float w = probs[t * topk + k];
      bf16x8 v = *(const bf16x8*)(permuted + p * h + i);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        acc[j] += w * bf2f((unsigned short)v[j]);
    }
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = (short)f2bf(acc[j]);
    *(bf16x8*)(out + t * h + i) = o;
  }
}

// one wave per permuted row
__global__ __launch_bounds__(MOE_BLOCK) void moe_combine_bwd_kernel(
    const unsigned short* __restrict__ dout,
    const unsigned short* __restrict__ permuted,
    const long* __restrict__ sort_idx, const float* __restrict__ probs,
    unsigned short* __restrict__ dpermuted, float* __restrict__ dprobs,
    long n_rows, int topk, int h) {
  const long p = (long)blockIdx.x * (MOE_BLOCK / WAVE) + threadIdx.x / WAVE;
  if (p >= n_rows) return;
  const int lane = threadIdx.x % WAVE;
  const long flat = sort_idx[p];
  const long t = flat / topk;
  const float w = probs[flat];
  float dot = 0.f;
  for (int i = lane * 8; i < h; i += WAVE * 8) {
    bf16x8 g = *(const bf16x8*)(dout + t * h + i);
    bf16x8 v = *(const bf16x8*)(permuted + p * h + i);
    bf16x8 dp;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gj = bf2f((unsigned short)g[j]);
      dot += gj * bf2f((unsigned short)v[j]);
      dp[j] = (short)f2bf(w * gj);
    }
    *(bf16x8*)(dpermuted + p * h + i) = dp;
  }
  dot = wave_reduce_sum(dot);
  if (lane == 0) dprobs[flat] = dot;
}

void launch_moe_combine_fwd(const void* permuted, const long* inv_pos,
                            const float* probs, void* out, long n_tokens,
                            int topk, int h, hipStream_t stream) {
  if (h % 8 != 0)
    throw std::runtime_error("moe_combine: h must be a multiple of 8");
  const int rpb = MOE_BLOCK / WAVE;
  hipLaunchKernelGGL(moe_combine_fwd_kernel,
                     dim3((unsigned)((n_tokens + rpb - 1) / rpb)),
                     dim3(MOE_BLOCK), 0, stream,
                     (const unsigned short*)permuted, inv_pos, probs,
                     (unsigned short*)out, n_tokens, topk, h);
  HIP_CHECK_LAUNCH();
}

void launch_moe_combine_bwd(const void* dout, const void* permuted,
                            const long* sort_idx, const float* probs,
                            void* dpermuted, float* dprobs, long n_rows,
                            int topk, int h, hipStream_t stream) {
  const int rpb = MOE_BLOCK / WAVE;
  hipLaunchKernelGGL(moe_combine_bwd_kernel,
                     dim3((unsigned)((n_rows + rpb - 1) / rpb)),
                     dim3(MOE_BLOCK), 0, stream,
                     (const unsigned short*)dout,
                     (const unsigned short*)permuted, sort_idx, probs,
                     (unsigned short*)dpermuted, dprobs, n_rows, topk, h);
  HIP_CHECK_LAUNCH();
}
