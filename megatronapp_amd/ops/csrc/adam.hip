// Fused AdamW over flat fp32 buffers — the "multi-tensor Adam" of the
// reference (apex multi_tensor_apply) collapsed to ONE
// contiguous span per DDP buffer by the buffer-aligned optimizer design
// (core/optimizer/distrib_optimizer.py).  Pure HBM-bound float4 stream.

#include "common.h"

#define BLOCK 256

__global__ void adamw_flat_kernel(float* __restrict__ p,
                                  const float* __restrict__ g,
                                  float* __restrict__ m, float* __restrict__ v,
                                  long n, float lr, float beta1, float beta2,
                                  float eps, float wd, float bc1, float bc2) {
  long i = ((long)blockIdx.x * BLOCK + threadIdx.x) * 4;
  const long stride = (long)gridDim.x * BLOCK * 4;
  const float decay = 1.f - lr * wd;
  const float step_size = lr / bc1;
  for (; i + 3 < n; i += stride) {
    float4v pv = *(float4v*)(p + i);
    float4v gv = *(const float4v*)(g + i);
    float4v mv = *(float4v*)(m + i);
    float4v vv = *(float4v*)(v + i);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float pj = pv[j] * decay;
      float mj = beta1 * mv[j] + (1.f - beta1) * gv[j];
      float vj = beta2 * vv[j] + (1.f - beta2) * gv[j] * gv[j];
      float denom = sqrtf(vj / bc2) + eps;
      pv[j] = pj - step_size * mj / denom;
      mv[j] = mj;
      vv[j] = vj;
    }
    *(float4v*)(p + i) = pv;
    *(float4v*)(m + i) = mv;
    *(float4v*)(v + i) = vv;
  }
  // tail
  if (blockIdx.x == 0 && threadIdx.x < 4) {
    long start = (n / 4) * 4;
    long k = start + threadIdx.x;
    if (k < n) {
      float pj = p[k] * decay;
      float mj = beta1 * m[k] + (1.f - beta1) * g[k];
      float vj = beta2 * v[k] + (1.f - beta2) * g[k] * g[k];
      p[k] = pj - step_size * mj / (sqrtf(vj / bc2) + eps);
      m[k] = mj;
      v[k] = vj;
    }
  }
}

// Range-table variant: identical update but weight decay is zeroed inside
// the no-wd ranges (biases/norms).  One launch per buffer instead of one
// per param slice; ~600 ranges resolved by a per-thread binary search on
// cached range tables (negligible next to the 24 B/element HBM stream).
__global__ void adamw_flat_ranged_kernel(
    float* __restrict__ p, const float* __restrict__ g, float* __restrict__ m,
    float* __restrict__ v, const long* __restrict__ nw_s,
    const long* __restrict__ nw_e, int n_ranges, long n, float lr, float beta1,
    float beta2, float eps, float wd, float bc1, float bc2) {
  long i = ((long)blockIdx.x * BLOCK + threadIdx.x) * 4;
  const long stride = (long)gridDim.x * BLOCK * 4;
  const float step_size = lr / bc1;
  for (; i + 3 < n; i += stride) {
    // no-wd iff i falls inside a [nw_s, nw_e) range (ranges are >=4-aligned
    // so one decision covers the float4)
    int lo = 0, hi = n_ranges;
    while (lo < hi) {
      int mid = (lo + hi) >> 1;
      if (nw_s[mid] <= i) lo = mid + 1; else hi = mid;
    }
    const bool no_wd = lo > 0 && i < nw_e[lo - 1];
    const float decay = no_wd ? 1.f : 1.f - lr * wd;
    float4v pv = *(float4v*)(p + i);
    float4v gv = *(const float4v*)(g + i);
    float4v mv = *(float4v*)(m + i);
    float4v vv = *(float4v*)(v + i);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float pj = pv[j] * decay;
      float mj = beta1 * mv[j] + (1.f - beta1) * gv[j];
      float vj = beta2 * vv[j] + (1.f - beta2) * gv[j] * gv[j];
      float denom = sqrtf(vj / bc2) + eps;
      pv[j] = pj - step_size * mj / denom;
      mv[j] = mj;
      vv[j] = vj;
    }
    *(float4v*)(p + i) = pv;
    *(float4v*)(m + i) = mv;
    *(float4v*)(v + i) = vv;
  }
  if (blockIdx.x == 0 && threadIdx.x < 4) {
    long start = (n / 4) * 4;
    long k = start + threadIdx.x;
    if (k < n) {
      int lo = 0, hi = n_ranges;
      while (lo < hi) {
        int mid = (lo + hi) >> 1;
        if (nw_s[mid] <= k) lo = mid + 1; else hi = mid;
      }
      const bool no_wd = lo > 0 && k < nw_e[lo - 1];
      const float decay = no_wd ? 1.f : 1.f - lr * wd;
      float pj = p[k] * decay;
      float mj = beta1 * m[k] + (1.f - beta1) * g[k];
      float vj = beta2 * v[k] + (1.f - beta2) * g[k] * g[k];
      p[k] = pj - step_size * mj / (sqrtf(vj / bc2) + eps);
      m[k] = mj;
      v[k] = vj;
    }
  }
}

void launch_adamw_flat_ranged(float* p, const float* g, float* m, float* v,
                              const long* nw_s, const long* nw_e, int n_ranges,
                              long n, float lr, float beta1, float beta2,
                              float eps, float wd, int step, hipStream_t s) {
  float bc1 = 1.f - powf(beta1, (float)step);
  float bc2 = 1.f - powf(beta2, (float)step);
  long blocks = (n / 4 + BLOCK - 1) / BLOCK;
  int grid = (int)(blocks < 2048 ? (blocks < 1 ? 1 : blocks) : 2048);
  hipLaunchKernelGGL(adamw_flat_ranged_kernel, dim3(grid), dim3(BLOCK), 0, s,
                     p, g, m, v, nw_s, nw_e, n_ranges, n, lr, beta1, beta2,
                     eps, wd, bc1, bc2);
  HIP_CHECK_LAUNCH();
}

void launch_adamw_flat(float* p, const float* g, float* m, float* v, long n,
                       float lr, float beta1, float beta2, float eps, float wd,
                       int step, hipStream_t s) {
  float bc1 = 1.f - powf(beta1, (float)step);
  float bc2 = 1.f - powf(beta2, (float)step);
  long blocks = (n / 4 + BLOCK - 1) / BLOCK;
  int grid = (int)(blocks < 2048 ? (blocks < 1 ? 1 : blocks) : 2048);
  hipLaunchKernelGGL(adamw_flat_kernel, dim3(grid), dim3(BLOCK), 0, s, p, g, m,
                     v, n, lr, beta1, beta2, eps, wd, bc1, bc2);
  HIP_CHECK_LAUNCH();
}
