// Fused RMSNorm / LayerNorm forward + backward for gfx950.
//
// Design (CDNA4 guide Appendix B, G13): memory-bound; bf16 loads as
// short8 (16 B/lane); one 256-thread block per row-group; fp32
// accumulation with wave shuffle + LDS block reduction; dw/db partials
// accumulated per block in registers over a grid-stride row loop, then
// one fp32 atomicAdd per column per block.
//
// Replaces: reference core/fusions/fused_layer_norm.py (apex kernels) and
// RMSNorm torch path.

#include "common.h"

#include <stdexcept>
#include <string>

#define BLOCK 256
// each thread handles VEC bf16 elements per row-chunk
#define VEC 8

// ---------------------------------------------------------------- RMSNorm fwd
__global__ void rmsnorm_fwd_kernel(const unsigned short* __restrict__ x,
                                   const unsigned short* __restrict__ w,
                                   unsigned short* __restrict__ y,
                                   float* __restrict__ invrms, int N, int H,
                                   float eps) {
  __shared__ float lds[BLOCK / WAVE];
  for (int row = blockIdx.x; row < N; row += gridDim.x) {
    const unsigned short* xr = x + (long)row * H;
    unsigned short* yr = y + (long)row * H;
    float ss = 0.f;
    for (int base = threadIdx.x * VEC; base < H; base += BLOCK * VEC) {
      short8v v = *(const short8v*)(xr + base);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float f = bf2f((unsigned short)v[j]);
        ss += f * f;
      }
    }
    ss = block_reduce_sum<BLOCK>(ss, lds);
    float r = rsqrtf(ss / H + eps);
    if (threadIdx.x == 0) invrms[row] = r;
    for (int base = threadIdx.x * VEC; base < H; base += BLOCK * VEC) {
      short8v v = *(const short8v*)(xr + base);
      short8v wv = *(const short8v*)(w + base);
      short8v o;
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        o[j] = (short)f2bf(bf2f((unsigned short)v[j]) * r *
                           bf2f((unsigned short)wv[j]));
      *(short8v*)(yr + base) = o;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------- RMSNorm bwd
// dw is accumulated into per-block partial buffers (dw_part[grid][H]) and
// reduced by norm_col_reduce_kernel — fp32 atomics across a 1024-deep grid
// serialized ~1000x on the same address (13% of step time in the first
// profile), partials + a tiny second kernel are contention-free.
template <int CHUNKS>
__global__ void rmsnorm_bwd_kernel(const unsigned short* __restrict__ dy,
                                   const unsigned short* __restrict__ x,
                                   const unsigned short* __restrict__ w,
                                   const float* __restrict__ invrms,
                                   unsigned short* __restrict__ dx,
                                   float* __restrict__ dw_part,
                                   const unsigned short* __restrict__ dres,
                                   int N, int H) {
  __shared__ float lds[BLOCK / WAVE];
  float dwacc[CHUNKS * VEC];
#pragma unroll
  for (int c = 0; c < CHUNKS * VEC; ++c) dwacc[c] = 0.f;

  for (int row = blockIdx.x; row < N; row += gridDim.x) {
    const unsigned short* xr = x + (long)row * H;
    const unsigned short* dyr = dy + (long)row * H;
    unsigned short* dxr = dx + (long)row * H;
    const float r = invrms[row];
    float dot = 0.f;
    // row tiles stay in registers between the two passes (<= 48 VGPRs
    // at CHUNKS=4): re-reading x/dy/w for the dx pass doubled the HBM
    // traffic of this kernel
    short8v xs[CHUNKS], dvs[CHUNKS], wvs[CHUNKS];
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      const int base = threadIdx.x * VEC + c * BLOCK * VEC;
      if (base >= H) break;
      short8v xv = xs[c] = *(const short8v*)(xr + base);
      short8v dv = dvs[c] = *(const short8v*)(dyr + base);
      short8v wv = wvs[c] = *(const short8v*)(w + base);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float xh = bf2f((unsigned short)xv[j]) * r;
        float dxh = bf2f((unsigned short)dv[j]) * bf2f((unsigned short)wv[j]);
        dot += dxh * xh;
        dwacc[c * VEC + j] += bf2f((unsigned short)dv[j]) * xh;
      }
    }
    dot = block_reduce_sum<BLOCK>(dot, lds) / H;
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      const int base = threadIdx.x * VEC + c * BLOCK * VEC;
      if (base >= H) break;
      short8v xv = xs[c];
      short8v dv = dvs[c];
      short8v wv = wvs[c];
      short8v o;
      // residual-join fusion: dx += the BDA residual's grad in the same
      // pass (saves a standalone [rows, H] add kernel per layer)
      short8v rv;
      if (dres != nullptr)
        rv = *(const short8v*)(dres + (long)row * H + base);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float xh = bf2f((unsigned short)xv[j]) * r;
        float dxh = bf2f((unsigned short)dv[j]) * bf2f((unsigned short)wv[j]);
        float add = (dres != nullptr) ? bf2f((unsigned short)rv[j]) : 0.f;
        o[j] = (short)f2bf(r * (dxh - xh * dot) + add);
      }
      *(short8v*)(dxr + base) = o;
    }
    __syncthreads();
  }
  float* dwp = dw_part + (long)blockIdx.x * H;
#pragma unroll
  for (int c = 0; c < CHUNKS; ++c) {
    const int base = threadIdx.x * VEC + c * BLOCK * VEC;
    if (base >= H) break;
#pragma unroll
    for (int j = 0; j < VEC; ++j) dwp[base + j] = dwacc[c * VEC + j];
  }
}

// column-reduce [G][H] fp32 partials into [H]: 2D grid (col chunk, g chunk)
// with one fp32 atomicAdd per (block, col).  out must be pre-zeroed, OR be
// the param's fp32 main_grad (gradient accumulation lands here directly).
#define GCHUNK 32
__global__ void norm_col_reduce_kernel(const float* __restrict__ part,
                                       float* __restrict__ out, int G, int H) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= H) return;
  const int g0 = blockIdx.y * GCHUNK;
  const int g1 = min(G, g0 + GCHUNK);
  float acc = 0.f;
  for (int g = g0; g < g1; ++g) acc += part[(long)g * H + col];
  atomicAdd(out + col, acc);
}

// -------------------------------------------------------------- LayerNorm fwd
__global__ void layernorm_fwd_kernel(const unsigned short* __restrict__ x,
                                     const unsigned short* __restrict__ w,
                                     const unsigned short* __restrict__ b,
                                     unsigned short* __restrict__ y,
                                     float* __restrict__ mean,
                                     float* __restrict__ invstd, int N, int H,
                                     float eps) {
  __shared__ float lds[2 * BLOCK / WAVE];
  for (int row = blockIdx.x; row < N; row += gridDim.x) {
    const unsigned short* xr = x + (long)row * H;
    unsigned short* yr = y + (long)row * H;
    float s = 0.f, ss = 0.f;
    for (int base = threadIdx.x * VEC; base < H; base += BLOCK * VEC) {
      short8v v = *(const short8v*)(xr + base);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float f = bf2f((unsigned short)v[j]);
        s += f;
        ss += f * f;
      }
    }
    float2 sss = block_reduce_sum2<BLOCK>(s, ss, lds);
    s = sss.x;
    ss = sss.y;
    float mu = s / H;
    float var = ss / H - mu * mu;
    float r = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      mean[row] = mu;
      invstd[row] = r;
    }
    for (int base = threadIdx.x * VEC; base < H; base += BLOCK * VEC) {
      short8v v = *(const short8v*)(xr + base);
      short8v wv = *(const short8v*)(w + base);
      short8v bv = *(const short8v*)(b + base);
      short8v o;
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        o[j] = (short)f2bf((bf2f((unsigned short)v[j]) - mu) * r *
                               bf2f((unsigned short)wv[j]) +
                           bf2f((unsigned short)bv[j]));
      *(short8v*)(yr + base) = o;
    }
    __syncthreads();
  }
}

// -------------------------------------------------------------- LayerNorm bwd
template <int CHUNKS>
__global__ void layernorm_bwd_kernel(const unsigned short* __restrict__ dy,
                                     const unsigned short* __restrict__ x,
                                     const unsigned short* __restrict__ w,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ invstd,
                                     unsigned short* __restrict__ dx,
                                     float* __restrict__ dw_part,
                                     float* __restrict__ db_part,
                                     const unsigned short* __restrict__ dres,
                                     int N, int H) {
  __shared__ float lds[2 * BLOCK / WAVE];
  float dwacc[CHUNKS * VEC];
  float dbacc[CHUNKS * VEC];
#pragma unroll
  for (int c = 0; c < CHUNKS * VEC; ++c) {
    dwacc[c] = 0.f;
    dbacc[c] = 0.f;
  }

  for (int row = blockIdx.x; row < N; row += gridDim.x) {
    const unsigned short* xr = x + (long)row * H;
    const unsigned short* dyr = dy + (long)row * H;
    unsigned short* dxr = dx + (long)row * H;
    const float mu = mean[row];
    const float r = invstd[row];
    float sum1 = 0.f, sum2 = 0.f;
    // register-resident row tiles between the passes (see rmsnorm_bwd)
    short8v xs[CHUNKS], dvs[CHUNKS], wvs[CHUNKS];
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      const int base = threadIdx.x * VEC + c * BLOCK * VEC;
      if (base >= H) break;
      short8v xv = xs[c] = *(const short8v*)(xr + base);
      short8v dv = dvs[c] = *(const short8v*)(dyr + base);
      short8v wv = wvs[c] = *(const short8v*)(w + base);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float xh = (bf2f((unsigned short)xv[j]) - mu) * r;
        float dyf = bf2f((unsigned short)dv[j]);
        float dxh = dyf * bf2f((unsigned short)wv[j]);
        sum1 += dxh;
        sum2 += dxh * xh;
        dwacc[c * VEC + j] += dyf * xh;
        dbacc[c * VEC + j] += dyf;
      }
    }
    float2 s12 = block_reduce_sum2<BLOCK>(sum1, sum2, lds);
    sum1 = s12.x / H;
    sum2 = s12.y / H;
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      const int base = threadIdx.x * VEC + c * BLOCK * VEC;
      if (base >= H) break;
      short8v xv = xs[c];
      short8v dv = dvs[c];
      short8v wv = wvs[c];
      short8v o;
      short8v rv;
      if (dres != nullptr)
        rv = *(const short8v*)(dres + (long)row * H + base);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float xh = (bf2f((unsigned short)xv[j]) - mu) * r;
        float dxh =
            bf2f((unsigned short)dv[j]) * bf2f((unsigned short)wv[j]);
        float add = (dres != nullptr) ? bf2f((unsigned short)rv[j]) : 0.f;
        o[j] = (short)f2bf(r * (dxh - sum1 - xh * sum2) + add);
      }
      *(short8v*)(dxr + base) = o;
    }
    __syncthreads();
  }
  float* dwp = dw_part + (long)blockIdx.x * H;
  float* dbp = db_part + (long)blockIdx.x * H;
#pragma unroll
  for (int c = 0; c < CHUNKS; ++c) {
    const int base = threadIdx.x * VEC + c * BLOCK * VEC;
    if (base >= H) break;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      dwp[base + j] = dwacc[c * VEC + j];
      dbp[base + j] = dbacc[c * VEC + j];
    }
  }
}

// ------------------------------------------------------------------ launchers
void launch_rmsnorm_fwd(const void* x, const void* w, void* y, float* invrms,
                        int N, int H, float eps, hipStream_t stream) {
  if (H % VEC != 0) throw std::runtime_error("H must be divisible by 8");
  int grid = N < 2048 ? N : 2048;
  hipLaunchKernelGGL(rmsnorm_fwd_kernel, dim3(grid), dim3(BLOCK), 0, stream,
                     (const unsigned short*)x, (const unsigned short*)w,
                     (unsigned short*)y, invrms, N, H, eps);
  HIP_CHECK_LAUNCH();
}

void launch_rmsnorm_bwd(const void* dy, const void* x, const void* w,
                        const float* invrms, void* dx, float* dw,
                        float* dw_part, const void* dres, int grid, int N,
                        int H, hipStream_t stream) {
  if (H % VEC != 0) throw std::runtime_error("H must be divisible by 8");
  const int chunks = (H + BLOCK * VEC - 1) / (BLOCK * VEC);
#define RMS_CASE(C)                                                           \
  case C:                                                                     \
    hipLaunchKernelGGL(rmsnorm_bwd_kernel<C>, dim3(grid), dim3(BLOCK), 0,     \
                       stream, (const unsigned short*)dy,                     \
                       (const unsigned short*)x, (const unsigned short*)w,    \
                       invrms, (unsigned short*)dx, dw_part,                  \
                       (const unsigned short*)dres, N, H);                    \
    break;
  switch (chunks) {
    RMS_CASE(1) RMS_CASE(2) RMS_CASE(3) RMS_CASE(4)
    default:
      throw std::runtime_error("H too large for rmsnorm bwd");
  }
#undef RMS_CASE
  HIP_CHECK_LAUNCH();
  hipLaunchKernelGGL(norm_col_reduce_kernel,
                     dim3((H + 255) / 256, (grid + GCHUNK - 1) / GCHUNK),
                     dim3(256), 0, stream, dw_part, dw, grid, H);
  HIP_CHECK_LAUNCH();
}

void launch_layernorm_fwd(const void* x, const void* w, const void* b, void* y,
                          float* mean, float* invstd, int N, int H, float eps,
                          hipStream_t stream) {
  if (H % VEC != 0) throw std::runtime_error("H must be divisible by 8");
  int grid = N < 2048 ? N : 2048;
  hipLaunchKernelGGL(layernorm_fwd_kernel, dim3(grid), dim3(BLOCK), 0, stream,
                     (const unsigned short*)x, (const unsigned short*)w,
                     (const unsigned short*)b, (unsigned short*)y, mean, invstd,
                     N, H, eps);
  HIP_CHECK_LAUNCH();
}

void launch_layernorm_bwd(const void* dy, const void* x, const void* w,
                          const float* mean, const float* invstd, void* dx,
                          float* dw, float* db, float* dw_part, float* db_part,
                          const void* dres, int grid, int N, int H,
                          hipStream_t stream) {
  if (H % VEC != 0) throw std::runtime_error("H must be divisible by 8");
  const int chunks = (H + BLOCK * VEC - 1) / (BLOCK * VEC);
#define LN_CASE(C)                                                            \
  case C:                                                                     \
    hipLaunchKernelGGL(layernorm_bwd_kernel<C>, dim3(grid), dim3(BLOCK), 0,   \
                       stream, (const unsigned short*)dy,                     \
                       (const unsigned short*)x, (const unsigned short*)w,    \
                       mean, invstd, (unsigned short*)dx, dw_part, db_part,   \
                       (const unsigned short*)dres, N, H);                    \
    break;
  switch (chunks) {
    LN_CASE(1) LN_CASE(2) LN_CASE(3) LN_CASE(4)
    default:
      throw std::runtime_error("H too large for layernorm bwd");
  }
#undef LN_CASE
  HIP_CHECK_LAUNCH();
  dim3 rg((H + 255) / 256, (grid + GCHUNK - 1) / GCHUNK);
  hipLaunchKernelGGL(norm_col_reduce_kernel, rg, dim3(256), 0, stream,
                     dw_part, dw, grid, H);
  hipLaunchKernelGGL(norm_col_reduce_kernel, rg, dim3(256), 0, stream,
                     db_part, db, grid, H);
  HIP_CHECK_LAUNCH();
}
