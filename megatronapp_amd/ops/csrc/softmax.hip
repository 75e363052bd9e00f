// Fused scale + mask + softmax kernels (bf16 I/O, fp32 math).
// Replaces apex scaled_{upper_triang_,}masked_softmax.
//
// Layout: scores [R, sk] where R = b*np*sq rows; causal variant derives
// the valid length from the row's q index.  One 256-thread block per
// row; grid-strided; online (max, sum) in fp32.

#include "common.h"

#include <stdexcept>

#define BLOCK 256
#define VEC 8

// ------------------------------------------------- causal (upper-triangular)
// Wave-per-row variants: 4 rows in flight per block, reductions are pure
// wave shuffles — no LDS, no __syncthreads.  Used when the row fits
// WAVE*VEC*NIT (sk <= 2048 at NIT 4).
template <int NIT>
__global__ void softmax_causal_fwd_wave_kernel(
    const unsigned short* __restrict__ x, unsigned short* __restrict__ y,
    long rows, int sq, int sk, float scale) {
  const int nwaves = BLOCK / WAVE;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  for (long row = (long)blockIdx.x * nwaves + wid; row < rows;
       row += (long)gridDim.x * nwaves) {
    const int q = (int)(row % sq);
    const int valid = q + 1 + (sk - sq);
    const unsigned short* xr = x + row * sk;
    unsigned short* yr = y + row * sk;

    short8v c[NIT];
    float m = -INFINITY;
#pragma unroll
    for (int it = 0; it < NIT; ++it) {
      const int base = it * WAVE * VEC + lane * VEC;
      if (base < valid) {
        c[it] = *(const short8v*)(xr + base);
#pragma unroll
        for (int j = 0; j < VEC; ++j)
          if (base + j < valid)
            m = fmaxf(m, bf2f((unsigned short)c[it][j]) * scale);
      }
    }
    m = wave_reduce_max(m);

    float pf[NIT][VEC];
    float sum = 0.f;
#pragma unroll
    for (int it = 0; it < NIT; ++it) {
      const int base = it * WAVE * VEC + lane * VEC;
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float p = (base + j < valid)
                      ? __expf(bf2f((unsigned short)c[it][j]) * scale - m)
                      : 0.f;
        pf[it][j] = p;
        sum += p;
      }
    }
    sum = wave_reduce_sum(sum);
    const float inv = 1.f / sum;
#pragma unroll
    for (int it = 0; it < NIT; ++it) {
      const int base = it * WAVE * VEC + lane * VEC;
      if (base < sk) {
        short8v o;
#pragma unroll
        for (int j = 0; j < VEC; ++j)
          o[j] = (short)f2bf(pf[it][j] * inv);
        *(short8v*)(yr + base) = o;
      }
    }
  }
}

template <int NIT, bool CAUSAL>
__global__ void softmax_bwd_wave_kernel(const unsigned short* __restrict__ dy,
                                        const unsigned short* __restrict__ p,
                                        unsigned short* __restrict__ dx,
                                        long rows, int sq, int sk,
                                        float scale) {
  const int nwaves = BLOCK / WAVE;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  for (long row = (long)blockIdx.x * nwaves + wid; row < rows;
       row += (long)gridDim.x * nwaves) {
    const int valid = CAUSAL ? (int)(row % sq) + 1 + (sk - sq) : sk;
    const unsigned short* dyr = dy + row * sk;
    const unsigned short* pr = p + row * sk;
    unsigned short* dxr = dx + row * sk;

    short8v cd[NIT], cp[NIT];
    float dot = 0.f;
#pragma unroll
    for (int it = 0; it < NIT; ++it) {
      const int base = it * WAVE * VEC + lane * VEC;
      if (base < valid) {
        cd[it] = *(const short8v*)(dyr + base);
        cp[it] = *(const short8v*)(pr + base);
#pragma unroll
        for (int j = 0; j < VEC; ++j)
          dot += bf2f((unsigned short)cd[it][j]) *
                 bf2f((unsigned short)cp[it][j]);
      }
    }
    dot = wave_reduce_sum(dot);
#pragma unroll
    for (int it = 0; it < NIT; ++it) {
      const int base = it * WAVE * VEC + lane * VEC;
      if (base >= sk) continue;
      short8v o;
      if (base < valid) {
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float pv = bf2f((unsigned short)cp[it][j]);
          o[j] = (short)f2bf(
              pv * (bf2f((unsigned short)cd[it][j]) - dot) * scale);
        }
      } else {
#pragma unroll
        for (int j = 0; j < VEC; ++j) o[j] = (short)0;
      }
      *(short8v*)(dxr + base) = o;
    }
  }
}

// Register-cached variants: one read of the row instead of three (fwd)
// or two (bwd), and causal rows only read their valid prefix.  Used when
// sk is a multiple of BLOCK*VEC and fits NIT<=4 iterations (sk <= 8192).
template <int NIT>
__global__ void softmax_causal_fwd_reg_kernel(
    const unsigned short* __restrict__ x, unsigned short* __restrict__ y,
    long rows, int sq, int sk, float scale) {
  __shared__ float lds[BLOCK / WAVE];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const int q = (int)(row % sq);
    const int valid = q + 1 + (sk - sq);
    const unsigned short* xr = x + row * sk;
    unsigned short* yr = y + row * sk;

    short8v c[NIT];
    float m = -INFINITY;
#pragma unroll
    for (int it = 0; it < NIT; ++it) {
      const int base = it * BLOCK * VEC + threadIdx.x * VEC;
      if (base < valid) {
        c[it] = *(const short8v*)(xr + base);
#pragma unroll
        for (int j = 0; j < VEC; ++j)
          if (base + j < valid)
            m = fmaxf(m, bf2f((unsigned short)c[it][j]) * scale);
      }
    }
    m = block_reduce_max<BLOCK>(m, lds);

    // p kept in fp32 regs between the sum and write passes: exp is the
    // dominant VALU cost and would otherwise run twice per element
    float pf[NIT][VEC];
    float sum = 0.f;
#pragma unroll
    for (int it = 0; it < NIT; ++it) {
      const int base = it * BLOCK * VEC + threadIdx.x * VEC;
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float p = (base + j < valid)
                      ? __expf(bf2f((unsigned short)c[it][j]) * scale - m)
                      : 0.f;
        pf[it][j] = p;
        sum += p;
      }
    }
    sum = block_reduce_sum<BLOCK>(sum, lds);
    const float inv = 1.f / sum;

#pragma unroll
    for (int it = 0; it < NIT; ++it) {
      const int base = it * BLOCK * VEC + threadIdx.x * VEC;
      short8v o;
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        o[j] = (short)f2bf(pf[it][j] * inv);
      *(short8v*)(yr + base) = o;
    }
    __syncthreads();
  }
}

template <int NIT, bool CAUSAL>
__global__ void softmax_bwd_reg_kernel(const unsigned short* __restrict__ dy,
                                       const unsigned short* __restrict__ p,
                                       unsigned short* __restrict__ dx,
                                       long rows, int sq, int sk,
                                       float scale) {
  __shared__ float lds[BLOCK / WAVE];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const int valid = CAUSAL ? (int)(row % sq) + 1 + (sk - sq) : sk;
    const unsigned short* dyr = dy + row * sk;
    const unsigned short* pr = p + row * sk;
    unsigned short* dxr = dx + row * sk;

    short8v cd[NIT], cp[NIT];
    float dot = 0.f;
#pragma unroll
    for (int it = 0; it < NIT; ++it) {
      const int base = it * BLOCK * VEC + threadIdx.x * VEC;
      if (base < valid) {
        cd[it] = *(const short8v*)(dyr + base);
        cp[it] = *(const short8v*)(pr + base);
#pragma unroll
        for (int j = 0; j < VEC; ++j)
          dot += bf2f((unsigned short)cd[it][j]) *
                 bf2f((unsigned short)cp[it][j]);
      }
    }
    dot = block_reduce_sum<BLOCK>(dot, lds);

#pragma unroll
    for (int it = 0; it < NIT; ++it) {
      const int base = it * BLOCK * VEC + threadIdx.x * VEC;
      short8v o;
      if (base < valid) {
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float pf = bf2f((unsigned short)cp[it][j]);
          o[j] = (short)f2bf(
              pf * (bf2f((unsigned short)cd[it][j]) - dot) * scale);
        }
      } else {
#pragma unroll
        for (int j = 0; j < VEC; ++j) o[j] = (short)0;
      }
      *(short8v*)(dxr + base) = o;
    }
    __syncthreads();
  }
}

__global__ void softmax_causal_fwd_kernel(const unsigned short* __restrict__ x,
                                          unsigned short* __restrict__ y,
                                          long rows, int sq, int sk,
                                          float scale) {
  __shared__ float lds[BLOCK / WAVE];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const int q = (int)(row % sq);
    const int valid = q + 1 + (sk - sq);  // causal with kv prefix
    const unsigned short* xr = x + row * sk;
    unsigned short* yr = y + row * sk;

    float m = -INFINITY;
    for (int base = threadIdx.x * VEC; base < sk; base += BLOCK * VEC) {
      short8v v = *(const short8v*)(xr + base);
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        if (base + j < valid) m = fmaxf(m, bf2f((unsigned short)v[j]) * scale);
    }
    m = block_reduce_max<BLOCK>(m, lds);

    float sum = 0.f;
    for (int base = threadIdx.x * VEC; base < sk; base += BLOCK * VEC) {
      short8v v = *(const short8v*)(xr + base);
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        if (base + j < valid)
          sum += __expf(bf2f((unsigned short)v[j]) * scale - m);
    }
    sum = block_reduce_sum<BLOCK>(sum, lds);
    const float inv = 1.f / sum;

    for (int base = threadIdx.x * VEC; base < sk; base += BLOCK * VEC) {
      short8v v = *(const short8v*)(xr + base);
      short8v o;
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        o[j] = (base + j < valid)
                   ? (short)f2bf(__expf(bf2f((unsigned short)v[j]) * scale - m) * inv)
                   : (short)0;
      *(short8v*)(yr + base) = o;
    }
    __syncthreads();
  }
}

// --------------------------------------------------------- generic bool mask
// mask: [mrows, sq, sk] uint8 (1 = masked out); broadcast over heads when
// mrows == b (mask row = row / (np*sq) * sq + q).
__global__ void softmax_masked_fwd_kernel(const unsigned short* __restrict__ x,
                                          const unsigned char* __restrict__ mask,
                                          unsigned short* __restrict__ y,
                                          long rows, int np, int sq, int sk,
                                          float scale) {
  __shared__ float lds[BLOCK / WAVE];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const int q = (int)(row % sq);
    const long bidx = row / ((long)np * sq);
    const unsigned char* mr =
        mask == nullptr ? nullptr : mask + (bidx * sq + q) * (long)sk;
    const unsigned short* xr = x + row * sk;
    unsigned short* yr = y + row * sk;

    float m = -INFINITY;
    for (int i = threadIdx.x; i < sk; i += BLOCK) {
      if (mr == nullptr || !mr[i])
        m = fmaxf(m, bf2f(xr[i]) * scale);
    }
    m = block_reduce_max<BLOCK>(m, lds);
    if (m == -INFINITY) m = 0.f;  // fully masked row

    float sum = 0.f;
    for (int i = threadIdx.x; i < sk; i += BLOCK) {
      if (mr == nullptr || !mr[i]) sum += __expf(bf2f(xr[i]) * scale - m);
    }
    sum = block_reduce_sum<BLOCK>(sum, lds);
    const float inv = sum > 0.f ? 1.f / sum : 0.f;

    for (int i = threadIdx.x; i < sk; i += BLOCK) {
      float p = (mr == nullptr || !mr[i])
                    ? __expf(bf2f(xr[i]) * scale - m) * inv
                    : 0.f;
      yr[i] = f2bf(p);
    }
    __syncthreads();
  }
}

// -------------------------------------------------------------------- bwd
// dx = p * (dy - sum(dy*p)) * scale     (rowwise)
__global__ void softmax_bwd_kernel(const unsigned short* __restrict__ dy,
                                   const unsigned short* __restrict__ p,
                                   unsigned short* __restrict__ dx, long rows,
                                   int sk, float scale) {
  __shared__ float lds[BLOCK / WAVE];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* dyr = dy + row * sk;
    const unsigned short* pr = p + row * sk;
    unsigned short* dxr = dx + row * sk;
    float dot = 0.f;
    for (int base = threadIdx.x * VEC; base < sk; base += BLOCK * VEC) {
      short8v d = *(const short8v*)(dyr + base);
      short8v pv = *(const short8v*)(pr + base);
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        dot += bf2f((unsigned short)d[j]) * bf2f((unsigned short)pv[j]);
    }
    dot = block_reduce_sum<BLOCK>(dot, lds);
    for (int base = threadIdx.x * VEC; base < sk; base += BLOCK * VEC) {
      short8v d = *(const short8v*)(dyr + base);
      short8v pv = *(const short8v*)(pr + base);
      short8v o;
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float pf = bf2f((unsigned short)pv[j]);
        o[j] = (short)f2bf(pf * (bf2f((unsigned short)d[j]) - dot) * scale);
      }
      *(short8v*)(dxr + base) = o;
    }
    __syncthreads();
  }
}

// ------------------------------------------------------------------ launchers
static int sm_grid(long rows) {
  return (int)(rows < 4096 ? (rows < 1 ? 1 : rows) : 4096);
}

void launch_softmax_causal_fwd(const void* x, void* y, long rows, int sq,
                               int sk, float scale, hipStream_t s) {
  if (sk % VEC != 0) throw std::runtime_error("sk must be divisible by 8");
  const int wspan = WAVE * VEC;
  if (sk % wspan == 0 && sk / wspan <= 4) {
#define FWD_WAVE(NIT)                                                         \
    hipLaunchKernelGGL((softmax_causal_fwd_wave_kernel<NIT>),                 \
                       dim3(sm_grid(rows / (BLOCK / WAVE) + 1)),              \
                       dim3(BLOCK), 0, s, (const unsigned short*)x,           \
                       (unsigned short*)y, rows, sq, sk, scale)
    switch (sk / wspan) {
      case 1: FWD_WAVE(1); break;
      case 2: FWD_WAVE(2); break;
      case 3: FWD_WAVE(3); break;
      default: FWD_WAVE(4); break;
    }
#undef FWD_WAVE
    HIP_CHECK_LAUNCH();
    return;
  }
  const int span = BLOCK * VEC;
  if (sk % span == 0 && sk / span <= 4) {
#define FWD_REG(NIT)                                                          \
    hipLaunchKernelGGL((softmax_causal_fwd_reg_kernel<NIT>),                  \
                       dim3(sm_grid(rows)), dim3(BLOCK), 0, s,                \
                       (const unsigned short*)x, (unsigned short*)y, rows,    \
                       sq, sk, scale)
    switch (sk / span) {
      case 1: FWD_REG(1); break;
      case 2: FWD_REG(2); break;
      case 3: FWD_REG(3); break;
      default: FWD_REG(4); break;
    }
#undef FWD_REG
    HIP_CHECK_LAUNCH();
    return;
  }
  hipLaunchKernelGGL(softmax_causal_fwd_kernel, dim3(sm_grid(rows)),
                     dim3(BLOCK), 0, s, (const unsigned short*)x,
                     (unsigned short*)y, rows, sq, sk, scale);
  HIP_CHECK_LAUNCH();
}

void launch_softmax_masked_fwd(const void* x, const void* mask, void* y,
                               long rows, int np, int sq, int sk, float scale,
                               hipStream_t s) {
  hipLaunchKernelGGL(softmax_masked_fwd_kernel, dim3(sm_grid(rows)),
                     dim3(BLOCK), 0, s, (const unsigned short*)x,
                     (const unsigned char*)mask, (unsigned short*)y, rows, np,
                     sq, sk, scale);
  HIP_CHECK_LAUNCH();
}

void launch_softmax_bwd_impl(const void* dy, const void* p, void* dx,
                             long rows, int sq, int sk, bool causal,
                             float scale, hipStream_t s) {
  if (sk % VEC != 0) throw std::runtime_error("sk must be divisible by 8");
  const int wspan = WAVE * VEC;
  if (sk % wspan == 0 && sk / wspan <= 4) {
#define BWD_WAVE(NIT, C)                                                      \
    hipLaunchKernelGGL((softmax_bwd_wave_kernel<NIT, C>),                     \
                       dim3(sm_grid(rows / (BLOCK / WAVE) + 1)),              \
                       dim3(BLOCK), 0, s, (const unsigned short*)dy,          \
                       (const unsigned short*)p, (unsigned short*)dx, rows,   \
                       sq, sk, scale)
    switch ((sk / wspan) * 2 + (causal ? 1 : 0)) {
      case 2 * 1 + 1: BWD_WAVE(1, true); break;
      case 2 * 1 + 0: BWD_WAVE(1, false); break;
      case 2 * 2 + 1: BWD_WAVE(2, true); break;
      case 2 * 2 + 0: BWD_WAVE(2, false); break;
      case 2 * 3 + 1: BWD_WAVE(3, true); break;
      case 2 * 3 + 0: BWD_WAVE(3, false); break;
      case 2 * 4 + 1: BWD_WAVE(4, true); break;
      default: BWD_WAVE(4, false); break;
    }
#undef BWD_WAVE
    HIP_CHECK_LAUNCH();
    return;
  }
  const int span = BLOCK * VEC;
  if (sk % span == 0 && sk / span <= 4) {
#define BWD_REG(NIT, C)                                                       \
    hipLaunchKernelGGL((softmax_bwd_reg_kernel<NIT, C>),                      \
                       dim3(sm_grid(rows)), dim3(BLOCK), 0, s,                \
                       (const unsigned short*)dy, (const unsigned short*)p,   \
                       (unsigned short*)dx, rows, sq, sk, scale)
    switch ((sk / span) * 2 + (causal ? 1 : 0)) {
      case 2 * 1 + 1: BWD_REG(1, true); break;
      case 2 * 1 + 0: BWD_REG(1, false); break;
      case 2 * 2 + 1: BWD_REG(2, true); break;
      case 2 * 2 + 0: BWD_REG(2, false); break;
      case 2 * 3 + 1: BWD_REG(3, true); break;
      case 2 * 3 + 0: BWD_REG(3, false); break;
      case 2 * 4 + 1: BWD_REG(4, true); break;
      default: BWD_REG(4, false); break;
    }
#undef BWD_REG
    HIP_CHECK_LAUNCH();
    return;
  }
  hipLaunchKernelGGL(softmax_bwd_kernel, dim3(sm_grid(rows)), dim3(BLOCK), 0,
                     s, (const unsigned short*)dy, (const unsigned short*)p,
                     (unsigned short*)dx, rows, sk, scale);
  HIP_CHECK_LAUNCH();
}

void launch_softmax_bwd(const void* dy, const void* p, void* dx, long rows,
                        int sk, float scale, hipStream_t s) {
  launch_softmax_bwd_impl(dy, p, dx, rows, sk, sk, false, scale, s);
}

void launch_softmax_causal_bwd(const void* dy, const void* p, void* dx,
                               long rows, int sq, int sk, float scale,
                               hipStream_t s) {
  launch_softmax_bwd_impl(dy, p, dx, rows, sq, sk, true, scale, s);
}
