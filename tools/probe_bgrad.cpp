// Probe: which hipblasLt BGRADA/BGRADB epilogue combinations does this
// build support for the wgrad shape, and what exactly do they reduce?
//
//   hipcc -O2 tools/probe_bgrad.cpp -lhipblaslt -o tools/probe_bgrad
//   gpurun -- tools/probe_bgrad
//
// wgrad wants: main_grad_rm[out,in] += grad_rm[rows,out]^T @ input_rm[rows,in]
// and dbias[out] = colsum(grad_rm).  Form1 puts grad as the (transposed)
// B operand; Form2 puts grad as the (normal) A operand but then D comes
// out transposed.

#include <hip/hip_runtime.h>
#include <hipblaslt/hipblaslt.h>

#include <cmath>
#include <cstdio>
#include <cstring>
#include <vector>

#define CHECK_HIP(x)                                            \
  if ((x) != hipSuccess) {                                      \
    printf("hip error %s at line %d\n", #x, __LINE__);          \
    return 1;                                                   \
  }

static unsigned short f2bf(float f) {
  unsigned int u;
  memcpy(&u, &f, 4);
  unsigned int lsb = (u >> 16) & 1;
  u += 0x7fff + lsb;
  return (unsigned short)(u >> 16);
}
static float bf2f(unsigned short h) {
  unsigned int u = (unsigned int)h << 16;
  float f;
  memcpy(&f, &u, 4);
  return f;
}

int main() {
  const int rows = 512, out = 384, in = 256;
  // deterministic small ints so bf16 is exact
  std::vector<unsigned short> hg(rows * out), hx(rows * in);
  std::vector<double> colsum(out, 0.0);
  for (int r = 0; r < rows; ++r)
    for (int o = 0; o < out; ++o) {
      float v = (float)((r * 7 + o * 3) % 5 - 2);
      hg[r * out + o] = f2bf(v);
      colsum[o] += v;
    }
  for (int r = 0; r < rows; ++r)
    for (int i = 0; i < in; ++i)
      hx[r * in + i] = f2bf((float)((r + 2 * i) % 3 - 1));

  void *dg, *dx, *dD32, *dD16, *dbias;
  CHECK_HIP(hipMalloc(&dg, hg.size() * 2));
  CHECK_HIP(hipMalloc(&dx, hx.size() * 2));
  CHECK_HIP(hipMalloc(&dD32, (size_t)out * in * 4));
  CHECK_HIP(hipMalloc(&dD16, (size_t)out * in * 2));
  CHECK_HIP(hipMalloc(&dbias, (size_t)out * 4));
  CHECK_HIP(hipMemcpy(dg, hg.data(), hg.size() * 2, hipMemcpyHostToDevice));
  CHECK_HIP(hipMemcpy(dx, hx.data(), hx.size() * 2, hipMemcpyHostToDevice));

  hipblasLtHandle_t handle;
  hipblasLtCreate(&handle);
  void* ws;
  CHECK_HIP(hipMalloc(&ws, 64 << 20));

  struct Cfg {
    const char* name;
    int form;               // 1: A=input opN, B=grad opT, BGRADB
                            // 2: A=grad opN, B=input opT, BGRADA
    hipDataType dtype;      // D dtype
    hipDataType btype;      // bias dtype
  };
  Cfg cfgs[] = {
      {"form1 BGRADB D=f32 bias=f32", 1, HIP_R_32F, HIP_R_32F},
      {"form1 BGRADB D=f32 bias=bf16", 1, HIP_R_32F, HIP_R_16BF},
      {"form1 BGRADB D=bf16 bias=f32", 1, HIP_R_16BF, HIP_R_32F},
      {"form1 BGRADB D=bf16 bias=bf16", 1, HIP_R_16BF, HIP_R_16BF},
      {"form2 BGRADA D=f32 bias=f32", 2, HIP_R_32F, HIP_R_32F},
      {"form2 BGRADA D=f32 bias=bf16", 2, HIP_R_32F, HIP_R_16BF},
      {"form2 BGRADA D=bf16 bias=f32", 2, HIP_R_16BF, HIP_R_32F},
  };

  for (const Cfg& cfg : cfgs) {
    int64_t M, N, K = rows;
    hipblasOperation_t opA = HIPBLAS_OP_N, opB = HIPBLAS_OP_T;
    const void *A, *B;
    int64_t lda, ldb;
    hipblasLtEpilogue_t epi;
    if (cfg.form == 1) {
      M = in; N = out;
      A = dx; lda = in;        // input_cm [in x rows]
      B = dg; ldb = out;       // grad stored [out x rows] cm, opT
      epi = HIPBLASLT_EPILOGUE_BGRADB;
    } else {
      M = out; N = in;
      A = dg; lda = out;       // grad_cm [out x rows]
      B = dx; ldb = in;        // input stored [in x rows] cm, opT
      epi = HIPBLASLT_EPILOGUE_BGRADA;
    }
    void* D = cfg.dtype == HIP_R_32F ? dD32 : dD16;

    hipblasLtMatmulDesc_t op{};
    hipblasLtMatmulDescCreate(&op, HIPBLAS_COMPUTE_32F, HIP_R_32F);
    hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_TRANSA, &opA,
                                    sizeof(opA));
    hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_TRANSB, &opB,
                                    sizeof(opB));
    hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_EPILOGUE, &epi,
                                    sizeof(epi));
    hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_BIAS_POINTER,
                                    &dbias, sizeof(void*));
    hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE,
                                    &cfg.btype, sizeof(cfg.btype));
    hipblasLtMatrixLayout_t la{}, lb{}, lc{};
    hipblasLtMatrixLayoutCreate(&la, HIP_R_16BF, M, K, M);
    hipblasLtMatrixLayoutCreate(&lb, HIP_R_16BF, N, K, N);
    hipblasLtMatrixLayoutCreate(&lc, cfg.dtype, M, N, M);

    hipblasLtMatmulPreference_t pref{};
    hipblasLtMatmulPreferenceCreate(&pref);
    size_t wsz = 64 << 20;
    hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &wsz, sizeof(wsz));
    hipblasLtMatmulHeuristicResult_t res[8];
    int found = 0;
    hipblasStatus_t hst = hipblasLtMatmulAlgoGetHeuristic(
        handle, op, la, lb, lc, lc, pref, 8, res, &found);
    hipblasLtMatmulPreferenceDestroy(pref);
    printf("%-32s heuristic: st=%d found=%d\n", cfg.name, (int)hst, found);
    if (hst == HIPBLAS_STATUS_SUCCESS && found > 0) {
      CHECK_HIP(hipMemset(dbias, 0, out * 4));
      CHECK_HIP(hipMemset(D, 0, (size_t)out * in *
                          (cfg.dtype == HIP_R_32F ? 4 : 2)));
      float alpha = 1.f, beta = 0.f;
      hipblasStatus_t mst = hipblasLtMatmul(
          handle, op, &alpha, A, la, B, lb, &beta, D, lc, D, lc, &res[0].algo,
          ws, 64 << 20, nullptr);
      CHECK_HIP(hipDeviceSynchronize());
      printf("    matmul st=%d", (int)mst);
      if (mst == HIPBLAS_STATUS_SUCCESS) {
        int blen = cfg.form == 1 ? out : out;  // expect len == grad cols
        if (cfg.btype == HIP_R_32F) {
          std::vector<float> hb(blen);
          hipMemcpy(hb.data(), dbias, blen * 4, hipMemcpyDeviceToHost);
          double err = 0;
          for (int o = 0; o < blen; ++o)
            err = fmax(err, fabs(hb[o] - colsum[o]));
          printf("  dbias[f32] max err vs colsum(grad): %g  (b0=%g exp=%g)",
                 err, hb[0], colsum[0]);
        } else {
          std::vector<unsigned short> hb(blen);
          hipMemcpy(hb.data(), dbias, blen * 2, hipMemcpyDeviceToHost);
          double err = 0;
          for (int o = 0; o < blen; ++o)
            err = fmax(err, fabs(bf2f(hb[o]) - colsum[o]));
          printf("  dbias[bf16] max err vs colsum(grad): %g (b0=%g exp=%g)",
                 err, bf2f(hb[0]), colsum[0]);
        }
      }
      printf("\n");
    }
    hipblasLtMatrixLayoutDestroy(la);
    hipblasLtMatrixLayoutDestroy(lb);
    hipblasLtMatrixLayoutDestroy(lc);
    hipblasLtMatmulDescDestroy(op);
  }
  printf("done\n");
  return 0;
}
