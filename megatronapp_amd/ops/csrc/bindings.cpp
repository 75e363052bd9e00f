// PyTorch bindings for the megatronapp_amd CDNA4 kernels.
// Built in-tree to megatronapp_amd/ops/_C.so (see ops/setup.py).

#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>

// launchers (norms.hip / elementwise.hip / rope.hip / softmax.hip / adam.hip /
// attention.hip)
void launch_rmsnorm_fwd(const void*, const void*, void*, float*, int, int,
                        float, hipStream_t);
void launch_rmsnorm_bwd(const void*, const void*, const void*, const float*,
                        void*, float*, float*, const void*, int, int, int,
                        hipStream_t);
void launch_layernorm_fwd(const void*, const void*, const void*, void*, float*,
                          float*, int, int, float, hipStream_t);
void launch_layernorm_bwd(const void*, const void*, const void*, const float*,
                          const float*, void*, float*, float*, float*, float*,
                          const void*, int, int, int, hipStream_t);
void launch_bias_add_residual(const void*, const void*, const void*, void*,
                              long, int, hipStream_t);
void launch_bias_gelu_fwd(const void*, const void*, void*, long, int,
                          hipStream_t);
void launch_bias_gelu_bwd(const void*, const void*, const void*, void*, long,
                          int, hipStream_t);
void launch_bias_swiglu_fwd(const void*, const void*, void*, long, int,
                            hipStream_t);
void launch_bias_swiglu_bwd(const void*, const void*, const void*, void*, long,
                            int, hipStream_t);
void launch_bias_geglu_fwd(const void*, const void*, void*, long, int,
                           hipStream_t);
void launch_bias_geglu_bwd(const void*, const void*, const void*, void*,
                           long, int, hipStream_t);
void launch_rope(const void*, const float*, const float*, void*, long, int,
                 int, bool, hipStream_t);
void launch_softmax_causal_fwd(const void*, void*, long, int, int, float,
                               hipStream_t);
void launch_softmax_masked_fwd(const void*, const void*, void*, long, int, int,
                               int, float, hipStream_t);
void launch_softmax_causal_bwd(const void*, const void*, void*, long, int,
                               int, float, hipStream_t);
void launch_softmax_bwd(const void*, const void*, void*, long, int, float,
                        hipStream_t);
void launch_adamw_flat_ranged(float*, const float*, void*, void*,
                              bool, void*, const long*, const long*,
                              int, long, float, float, float, float,
                              float, int, hipStream_t);
void launch_ce_rowmax(const void*, float*, long, int, hipStream_t);
void launch_ce_fwd(const void*, const float*, const int*, float*, float*,
                   long, int, hipStream_t);
void launch_ce_bwd(const void*, const float*, const float*, const int*,
                   const float*, void*, long, int, hipStream_t);
void launch_selective_scan_fwd(const void*, const void*, const float*,
                               const void*, const void*, const float*, float*,
                               void*, int, int, int, int, hipStream_t);
void launch_adamw_flat(float*, const float*, void*, void*, bool,
                       void*, long, float, float, float, float,
                       float, int, hipStream_t);
void wgrad_accum(torch::Tensor, torch::Tensor, torch::Tensor);
bool wgrad_accum_bgrad(torch::Tensor, torch::Tensor, torch::Tensor,
                       torch::Tensor);
torch::Tensor gemm_nt(torch::Tensor, torch::Tensor);
torch::Tensor gemm_nn(torch::Tensor, torch::Tensor);
int colsum_grid_y(long, int);
void launch_colsum_accum(const void*, float*, float*, int, long, int,
                         hipStream_t);
void launch_embedding_bwd_accum(const void*, const int*, float*, long, int,
                                hipStream_t);
void launch_transpose_quant_e4m3(const void*, void*, unsigned int*,
                                 float*, long, long, hipStream_t);
void launch_quant_rows_e4m3(const void*, void*, float*, long, int,
                            hipStream_t);
void launch_bias_dropout_add_fwd(const void*, const void*, const void*,
                                 void*, unsigned char*, long, int, float,
                                 unsigned long long, hipStream_t);
void launch_dropout_bwd(const void*, const unsigned char*, void*, long,
                        float, hipStream_t);
void launch_moe_combine_fwd(const void*, const long*, const float*, void*,
                            long, int, int, hipStream_t);
void launch_moe_combine_bwd(const void*, const void*, const long*,
                            const float*, void*, float*, long, int, int,
                            hipStream_t);
void launch_attn_fwd2(const void*, const void*, const void*, void*, float*,
                      int, int, int, int, int, int, float, bool,
                      const long*, const long*, const long*, hipStream_t);
void launch_attn_bwd2(const void*, const void*, const void*, const void*,
                      const void*, const float*, float*, void*, void*, void*,
                      int, int, int, int, int, int, float, bool, const long*,
                      const long*, const long*, const long*, const long*,
                      const long*, hipStream_t);

// The round-2 kernels take (seq, batch, head) element strides so the
// strided QKV-split views feed them without .contiguous() copies; d
// must be unit-stride.
static bool attn_strides(const torch::Tensor& t, long* out) {
  if (t.stride(3) != 1) return false;
  out[0] = (long)t.stride(0);
  out[1] = (long)t.stride(1);
  out[2] = (long)t.stride(2);
  return true;
}
void launch_attn_fwd_t(const void*, const void*, const void*, void*, float*,
                       int, int, int, int, int, int, float, bool,
                       hipStream_t);
void launch_attn_fwd(const void*, const void*, const void*, void*, float*,
                     int, int, int, int, int, int, float, bool, hipStream_t);
void launch_attn_fwd_ablate(const void*, const void*, const void*, void*,
                            float*, int, int, int, int, int, int, float, int,
                            hipStream_t);
void launch_attn_bwd(const void*, const void*, const void*, const void*,
                     const void*, const float*, float*, void*, void*, void*,
                     int, int, int, int, int, int, float, bool, hipStream_t);

namespace {

hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// attention inputs may be strided views (d contiguous); no contiguity check
void check_bf16_any(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
}

// ------------------------------------------------------------------- norms
std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                                       double eps) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  const int H = (int)x.size(-1);
  const long N = x.numel() / H;
  auto y = torch::empty_like(x);
  auto invrms = torch::empty({N}, x.options().dtype(torch::kFloat32));
  launch_rmsnorm_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                     invrms.data_ptr<float>(), (int)N, H, (float)eps,
                     cur_stream());
  return {y, invrms};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor invrms,
                                       c10::optional<torch::Tensor> mg_w,
                                       c10::optional<torch::Tensor> dres) {
  check_bf16(dy, "dy");
  check_bf16(x, "x");
  const int H = (int)x.size(-1);
  const long N = x.numel() / H;
  auto dx = torch::empty_like(x);
  auto dw = mg_w.has_value()
                ? *mg_w
                : torch::zeros({H}, x.options().dtype(torch::kFloat32));
  // 2048 blocks = up to 8/CU: a block sits at two block-wide barrier
  // reductions per row, so a 512-block launch left the CUs idle
  const int grid = (int)std::min<long>(N, 2048);
  auto dw_part = torch::empty({grid, H}, x.options().dtype(torch::kFloat32));
  const void* dres_p = nullptr;
  torch::Tensor dres_c;
  if (dres.has_value()) {
    check_bf16(*dres, "dres");
    dres_c = dres->contiguous();
    dres_p = dres_c.data_ptr();
  }
  launch_rmsnorm_bwd(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                     invrms.data_ptr<float>(), dx.data_ptr(),
                     dw.data_ptr<float>(), dw_part.data_ptr<float>(), dres_p,
                     grid, (int)N, H, cur_stream());
  return {dx, dw};
}

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps) {
  check_bf16(x, "x");
  const int H = (int)x.size(-1);
  const long N = x.numel() / H;
  auto y = torch::empty_like(x);
  auto mean = torch::empty({N}, x.options().dtype(torch::kFloat32));
  auto invstd = torch::empty({N}, x.options().dtype(torch::kFloat32));
  launch_layernorm_fwd(x.data_ptr(), w.data_ptr(), b.data_ptr(), y.data_ptr(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       (int)N, H, (float)eps, cur_stream());
  return {y, mean, invstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor invstd,
                                         c10::optional<torch::Tensor> mg_w,
                                         c10::optional<torch::Tensor> mg_b,
                                         c10::optional<torch::Tensor> dres) {
  check_bf16(dy, "dy");
  const int H = (int)x.size(-1);
  const long N = x.numel() / H;
  auto dx = torch::empty_like(x);
  auto dw = mg_w.has_value()
                ? *mg_w
                : torch::zeros({H}, x.options().dtype(torch::kFloat32));
  auto db = mg_b.has_value()
                ? *mg_b
                : torch::zeros({H}, x.options().dtype(torch::kFloat32));
  // 2048 blocks = up to 8/CU: a block sits at two block-wide barrier
  // reductions per row, so a 512-block launch left the CUs idle
  const int grid = (int)std::min<long>(N, 2048);
  auto part = torch::empty({2, grid, H}, x.options().dtype(torch::kFloat32));
  const void* dres_p = nullptr;
  torch::Tensor dres_c;
  if (dres.has_value()) {
    check_bf16(*dres, "dres");
    dres_c = dres->contiguous();
    dres_p = dres_c.data_ptr();
  }
  launch_layernorm_bwd(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       dx.data_ptr(), dw.data_ptr<float>(),
                       db.data_ptr<float>(), part[0].data_ptr<float>(),
                       part[1].data_ptr<float>(), dres_p, grid, (int)N, H,
                       cur_stream());
  return {dx, dw, db};
}

// -------------------------------------------------------------- elementwise
torch::Tensor bias_gelu_fwd(torch::Tensor x, c10::optional<torch::Tensor> bias) {
  check_bf16(x, "x");
  auto y = torch::empty_like(x);
  const int F = (int)x.size(-1);
  TORCH_CHECK(x.numel() % 8 == 0 && (!bias.has_value() || F % 8 == 0),
              "bias_gelu: numel and F must be multiples of 8");
  launch_bias_gelu_fwd(x.data_ptr(),
                       bias.has_value() ? bias->data_ptr() : nullptr,
                       y.data_ptr(), x.numel(), F, cur_stream());
  return y;
}

torch::Tensor bias_add_residual(torch::Tensor x, torch::Tensor bias,
                                torch::Tensor res) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && res.is_contiguous() &&
              bias.is_contiguous());
  TORCH_CHECK(x.sizes() == res.sizes());
  int F = bias.numel();
  TORCH_CHECK(x.size(-1) == F && F % 8 == 0);
  auto y = torch::empty_like(x);
  launch_bias_add_residual(x.data_ptr(), bias.data_ptr(), res.data_ptr(),
                           y.data_ptr(), x.numel(), F,
                           at::cuda::getCurrentCUDAStream());
  return y;
}

torch::Tensor bias_gelu_bwd(torch::Tensor dy, torch::Tensor x,
                            c10::optional<torch::Tensor> bias) {
  check_bf16(dy, "dy");
  auto dx = torch::empty_like(x);
  const int F = (int)x.size(-1);
  TORCH_CHECK(x.numel() % 8 == 0 && (!bias.has_value() || F % 8 == 0),
              "bias_gelu: numel and F must be multiples of 8");
  launch_bias_gelu_bwd(dy.data_ptr(), x.data_ptr(),
                       bias.has_value() ? bias->data_ptr() : nullptr,
                       dx.data_ptr(), x.numel(), F, cur_stream());
  return dx;
}

torch::Tensor bias_swiglu_fwd(torch::Tensor x, c10::optional<torch::Tensor> bias) {
  check_bf16(x, "x");
  const int F2 = (int)x.size(-1);
  TORCH_CHECK(F2 % 2 == 0, "last dim must be even (gated)");
  const int F = F2 / 2;
  const long N = x.numel() / F2;
  auto sizes = x.sizes().vec();
  sizes.back() = F;
  auto y = torch::empty(sizes, x.options());
  launch_bias_swiglu_fwd(x.data_ptr(),
                         bias.has_value() ? bias->data_ptr() : nullptr,
                         y.data_ptr(), N, F, cur_stream());
  return y;
}

torch::Tensor bias_swiglu_bwd(torch::Tensor dy, torch::Tensor x,
                              c10::optional<torch::Tensor> bias) {
  check_bf16(dy, "dy");
  const int F2 = (int)x.size(-1);
  const int F = F2 / 2;
  const long N = x.numel() / F2;
  auto dx = torch::empty_like(x);
  launch_bias_swiglu_bwd(dy.data_ptr(), x.data_ptr(),
                         bias.has_value() ? bias->data_ptr() : nullptr,
                         dx.data_ptr(), N, F, cur_stream());
  return dx;
}

torch::Tensor bias_geglu_fwd(torch::Tensor x, c10::optional<torch::Tensor> bias) {
  check_bf16(x, "x");
  const int F2 = (int)x.size(-1);
  TORCH_CHECK(F2 % 2 == 0, "last dim must be even (gated)");
  const int F = F2 / 2;
  const long N = x.numel() / F2;
  auto sizes = x.sizes().vec();
  sizes.back() = F;
  auto y = torch::empty(sizes, x.options());
  launch_bias_geglu_fwd(x.data_ptr(),
                         bias.has_value() ? bias->data_ptr() : nullptr,
                         y.data_ptr(), N, F, cur_stream());
  return y;
}

torch::Tensor bias_geglu_bwd(torch::Tensor dy, torch::Tensor x,
                              c10::optional<torch::Tensor> bias) {
  check_bf16(dy, "dy");
  const int F2 = (int)x.size(-1);
  const int F = F2 / 2;
  const long N = x.numel() / F2;
  auto dx = torch::empty_like(x);
  launch_bias_geglu_bwd(dy.data_ptr(), x.data_ptr(),
                         bias.has_value() ? bias->data_ptr() : nullptr,
                         dx.data_ptr(), N, F, cur_stream());
  return dx;
}

// --------------------------------------------------------------------- rope
torch::Tensor rope_apply(torch::Tensor t, torch::Tensor cs, torch::Tensor sn,
                         bool bwd) {
  check_bf16(t, "t");
  TORCH_CHECK(cs.scalar_type() == torch::kFloat32, "cos table must be fp32");
  const int d = (int)t.size(-1);
  const long rows = t.numel() / d;
  // cs/sn come in as [s, 1, 1, d]; rows per seq position = b*nh
  const long s = cs.numel() / d;
  const int bnh = (int)(rows / s);
  auto out = torch::empty_like(t);
  launch_rope(t.data_ptr(), cs.data_ptr<float>(), sn.data_ptr<float>(),
              out.data_ptr(), rows, bnh, d, bwd, cur_stream());
  return out;
}

torch::Tensor rope_fwd(torch::Tensor t, torch::Tensor cs, torch::Tensor sn) {
  return rope_apply(t, cs.contiguous(), sn.contiguous(), false);
}

torch::Tensor rope_bwd(torch::Tensor dy, torch::Tensor cs, torch::Tensor sn) {
  return rope_apply(dy, cs.contiguous(), sn.contiguous(), true);
}

// ------------------------------------------------------------------ softmax
torch::Tensor scaled_softmax_fwd(torch::Tensor x, double scale) {
  // no-mask row softmax (reference scaled_softmax_cuda): the masked
  // kernel with a null mask pointer
  check_bf16(x, "x");
  const int sk = (int)x.size(-1);
  const int sq = (int)x.size(-2);
  const long rows = x.numel() / sk;
  auto y = torch::empty_like(x);
  launch_softmax_masked_fwd(x.data_ptr(), nullptr, y.data_ptr(), rows,
                            1, sq, sk, (float)scale, cur_stream());
  return y;
}

torch::Tensor scaled_upper_triang_masked_softmax_fwd(torch::Tensor x,
                                                     double scale) {
  check_bf16(x, "x");
  const int sk = (int)x.size(-1);
  const int sq = (int)x.size(-2);
  const long rows = x.numel() / sk;
  auto y = torch::empty_like(x);
  launch_softmax_causal_fwd(x.data_ptr(), y.data_ptr(), rows, sq, sk,
                            (float)scale, cur_stream());
  return y;
}

torch::Tensor scaled_masked_softmax_fwd(torch::Tensor x,
                                        c10::optional<torch::Tensor> mask,
                                        double scale) {
  check_bf16(x, "x");
  // x: [b, np, sq, sk]
  const int sk = (int)x.size(-1);
  const int sq = (int)x.size(-2);
  const int np = (int)x.size(1);
  const long rows = x.numel() / sk;
  auto y = torch::empty_like(x);
  const void* mptr = nullptr;
  torch::Tensor m8;
  if (mask.has_value()) {
    m8 = mask->to(torch::kUInt8).contiguous();
    mptr = m8.data_ptr();
  }
  launch_softmax_masked_fwd(x.data_ptr(), mptr, y.data_ptr(), rows, np, sq, sk,
                            (float)scale, cur_stream());
  return y;
}

torch::Tensor scaled_softmax_bwd(torch::Tensor dy, torch::Tensor p,
                                 double scale) {
  check_bf16(dy, "dy");
  const int sk = (int)dy.size(-1);
  const long rows = dy.numel() / sk;
  auto dx = torch::empty_like(dy);
  launch_softmax_bwd(dy.data_ptr(), p.data_ptr(), dx.data_ptr(), rows, sk,
                     (float)scale, cur_stream());
  return dx;
}

torch::Tensor scaled_upper_triang_masked_softmax_bwd(torch::Tensor dy,
                                                     torch::Tensor p,
                                                     double scale) {
  check_bf16(dy, "dy");
  const int sk = (int)dy.size(-1);
  const int sq = (int)dy.size(-2);
  const long rows = dy.numel() / sk;
  auto dx = torch::empty_like(dy);
  launch_softmax_causal_bwd(dy.data_ptr(), p.data_ptr(), dx.data_ptr(), rows,
                            sq, sk, (float)scale, cur_stream());
  return dx;
}

torch::Tensor colsum_accum(torch::Tensor dy, torch::Tensor out) {
  check_bf16(dy, "dy");
  TORCH_CHECK(out.scalar_type() == torch::kFloat32 && out.is_contiguous());
  const int F = (int)dy.size(-1);
  const long R = dy.numel() / F;
  const int gy = colsum_grid_y(R, F);
  torch::Tensor part;
  float* part_p = nullptr;
  if (gy > 0) {
    part = torch::empty({(long)gy, (long)F},
                        out.options().dtype(torch::kFloat32));
    part_p = part.data_ptr<float>();
  }
  launch_colsum_accum(dy.data_ptr(), out.data_ptr<float>(), part_p, gy, R, F,
                      cur_stream());
  return out;
}

void embedding_bwd_accum(torch::Tensor dy, torch::Tensor tokens,
                         torch::Tensor main_grad) {
  check_bf16(dy, "dy");
  TORCH_CHECK(tokens.scalar_type() == torch::kInt32 && tokens.is_contiguous());
  TORCH_CHECK(main_grad.scalar_type() == torch::kFloat32 &&
              main_grad.is_contiguous());
  const int H = (int)dy.size(-1);
  const long ntok = dy.numel() / H;
  TORCH_CHECK(tokens.numel() == ntok);
  launch_embedding_bwd_accum(dy.data_ptr(), tokens.data_ptr<int>(),
                             main_grad.data_ptr<float>(), ntok, H,
                             cur_stream());
}

// ------------------------------------------------------------- dropout
std::vector<torch::Tensor> bias_dropout_add_fwd(
    torch::Tensor x, torch::Tensor bias, torch::Tensor residual, double p,
    long seed) {
  check_bf16(x, "x");
  check_bf16(residual, "residual");
  auto out = torch::empty_like(x);
  auto mask = torch::empty(x.sizes(), x.options().dtype(torch::kUInt8));
  const long n = x.numel();
  const int F = (int)x.size(-1);
  launch_bias_dropout_add_fwd(
      x.data_ptr(), bias.defined() ? bias.data_ptr() : nullptr,
      residual.data_ptr(), out.data_ptr(), mask.data_ptr<unsigned char>(),
      n, F, (float)p, (unsigned long long)seed, cur_stream());
  return {out, mask};
}

torch::Tensor dropout_bwd(torch::Tensor dy, torch::Tensor mask, double p) {
  check_bf16(dy, "dy");
  auto dx = torch::empty_like(dy);
  launch_dropout_bwd(dy.contiguous().data_ptr(),
                     mask.data_ptr<unsigned char>(), dx.data_ptr(),
                     dy.numel(), (float)p, cur_stream());
  return dx;
}

// ---------------------------------------------------------------- moe
torch::Tensor moe_combine_fwd(torch::Tensor permuted, torch::Tensor inv_pos,
                              torch::Tensor probs, long n_tokens) {
  check_bf16(permuted, "permuted");
  const int h = (int)permuted.size(-1);
  const int topk = (int)(permuted.size(0) / n_tokens);
  auto out = torch::empty({n_tokens, (long)h}, permuted.options());
  launch_moe_combine_fwd(permuted.data_ptr(), inv_pos.data_ptr<long>(),
                         probs.data_ptr<float>(), out.data_ptr(), n_tokens,
                         topk, h, cur_stream());
  return out;
}

std::vector<torch::Tensor> moe_combine_bwd(torch::Tensor dout,
                                           torch::Tensor permuted,
                                           torch::Tensor sort_idx,
                                           torch::Tensor probs, long topk) {
  check_bf16(dout, "dout");
  check_bf16(permuted, "permuted");
  const int h = (int)permuted.size(-1);
  const long n_rows = permuted.size(0);
  auto dpermuted = torch::empty_like(permuted);
  auto dprobs = torch::empty({n_rows}, probs.options());
  launch_moe_combine_bwd(dout.contiguous().data_ptr(), permuted.data_ptr(),
                         sort_idx.data_ptr<long>(), probs.data_ptr<float>(),
                         dpermuted.data_ptr(), dprobs.data_ptr<float>(),
                         n_rows, (int)topk, h, cur_stream());
  return {dpermuted, dprobs};
}

// ---------------------------------------------------------------- fp8 quant
// [R, C] bf16 -> ([C, R] e4m3, [1] f32 per-tensor scale): wgrad operands
// (fp8 MFMA wants the token dim innermost on both sides)
std::vector<torch::Tensor> quantize_transpose_e4m3(torch::Tensor x) {
  check_bf16(x, "x");
  TORCH_CHECK(x.dim() == 2, "expects 2D");
  const long R = x.size(0), C = x.size(1);
  auto q = torch::empty({C, R}, x.options().dtype(torch::kFloat8_e4m3fn));
  auto s = torch::empty({1}, x.options().dtype(torch::kFloat32));
  auto amax = torch::zeros({1}, x.options().dtype(torch::kInt32));
  launch_transpose_quant_e4m3(x.data_ptr(), q.data_ptr(),
                              (unsigned int*)amax.data_ptr(),
                              s.data_ptr<float>(), R, C, cur_stream());
  return {q, s};
}

std::vector<torch::Tensor> quantize_rows_e4m3(torch::Tensor x) {
  check_bf16(x, "x");
  auto x2 = x.reshape({-1, x.size(-1)}).contiguous();
  const long M = x2.size(0);
  const int K = (int)x2.size(1);
  auto q = torch::empty({M, (long)K},
                        x.options().dtype(torch::kFloat8_e4m3fn));
  auto s = torch::empty({M, 1}, x.options().dtype(torch::kFloat32));
  launch_quant_rows_e4m3(x2.data_ptr(), q.data_ptr(), s.data_ptr<float>(),
                         M, K, cur_stream());
  return {q, s};
}

// ---------------------------------------------------------------- attention
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, double scale,
                                    bool causal) {
  check_bf16_any(q, "q");
  check_bf16_any(k, "k");
  check_bf16_any(v, "v");
  const int sq = (int)q.size(0), b = (int)q.size(1), nh = (int)q.size(2),
            d = (int)q.size(3);
  const int sk = (int)k.size(0), ng = (int)k.size(2);
  auto o = torch::empty({sq, b, nh, d}, q.options());
  auto lse = torch::empty({b, nh, sq}, q.options().dtype(torch::kFloat32));
  long qs[3], ks[3], vs[3];
  if ((d == 128 || d == 64) && sq % 256 == 0 && sk % 64 == 0 &&
      attn_strides(q, qs) &&
      attn_strides(k, ks) && attn_strides(v, vs)) {
    launch_attn_fwd2(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                     lse.data_ptr<float>(), sq, sk, b, nh, ng, d,
                     (float)scale, causal, qs, ks, vs, cur_stream());
  } else {
    auto qc = q.contiguous(), kc = k.contiguous(), vc = v.contiguous();
    launch_attn_fwd(qc.data_ptr(), kc.data_ptr(), vc.data_ptr(), o.data_ptr(),
                    lse.data_ptr<float>(), sq, sk, b, nh, ng, d, (float)scale,
                    causal, cur_stream());
  }
  return {o, lse};
}

std::vector<torch::Tensor> attn_fwd_t(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, double scale,
                                    bool causal) {
  check_bf16_any(q, "q");
  check_bf16_any(k, "k");
  check_bf16_any(v, "v");
  const int sq = (int)q.size(0), b = (int)q.size(1), nh = (int)q.size(2),
            d = (int)q.size(3);
  const int sk = (int)k.size(0), ng = (int)k.size(2);
  auto o = torch::empty_like(q);
  auto lse = torch::empty({b, nh, sq}, q.options().dtype(torch::kFloat32));
  launch_attn_fwd_t(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                  lse.data_ptr<float>(), sq, sk, b, nh, ng, d, (float)scale,
                  causal, cur_stream());
  return {o, lse};
}

std::vector<torch::Tensor> attn_fwd2(torch::Tensor q, torch::Tensor k,
                                     torch::Tensor v, double scale,
                                     bool causal) {
  check_bf16_any(q, "q");
  check_bf16_any(k, "k");
  check_bf16_any(v, "v");
  const int sq = (int)q.size(0), b = (int)q.size(1), nh = (int)q.size(2),
            d = (int)q.size(3);
  const int sk = (int)k.size(0), ng = (int)k.size(2);
  auto o = torch::empty({sq, b, nh, d}, q.options());
  auto lse = torch::empty({b, nh, sq}, q.options().dtype(torch::kFloat32));
  long qs[3], ks[3], vs[3];
  TORCH_CHECK(attn_strides(q, qs) && attn_strides(k, ks) &&
              attn_strides(v, vs), "attn_fwd2: d must be contiguous");
  launch_attn_fwd2(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                   lse.data_ptr<float>(), sq, sk, b, nh, ng, d, (float)scale,
                   causal, qs, ks, vs, cur_stream());
  return {o, lse};
}

std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q,
                                    torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse,
                                    double scale, bool causal) {
  check_bf16_any(dout, "dout");
  check_bf16_any(q, "q");
  check_bf16_any(k, "k");
  check_bf16_any(v, "v");
  const int sq = (int)q.size(0), b = (int)q.size(1), nh = (int)q.size(2),
            d = (int)q.size(3);
  const int sk = (int)k.size(0), ng = (int)k.size(2);
  // q/k/v may be gapped views of the fused QKV GEMM output; the
  // gradients are plain contiguous [s, b, h, d] tensors
  auto dq = torch::empty({(long)sq, b, nh, d}, q.options());
  auto dk = torch::empty({(long)sk, b, ng, d}, q.options());
  auto dv = torch::empty({(long)sk, b, ng, d}, q.options());
  auto drow = torch::empty({(long)sq * b * nh},
                           q.options().dtype(torch::kFloat32));
  long qs[3], ks[3], vs[3];
  auto doc = dout.contiguous();
  if ((d == 128 || d == 64) && sq % 256 == 0 && sk % 256 == 0 &&
      attn_strides(q, qs) &&
      attn_strides(k, ks) && attn_strides(v, vs)) {
    const long dqs[3] = {(long)b * nh * d, (long)nh * d, (long)d};
    const long dks[3] = {(long)b * ng * d, (long)ng * d, (long)d};
    launch_attn_bwd2(doc.data_ptr(), q.data_ptr(), k.data_ptr(),
                     v.data_ptr(), o.data_ptr(), lse.data_ptr<float>(),
                     drow.data_ptr<float>(), dq.data_ptr(), dk.data_ptr(),
                     dv.data_ptr(), sq, sk, b, nh, ng, d, (float)scale,
                     causal, qs, ks, vs, dqs, dks, dks, cur_stream());
  } else {
    auto qc = q.contiguous(), kc = k.contiguous(), vc = v.contiguous();
    launch_attn_bwd(doc.data_ptr(), qc.data_ptr(), kc.data_ptr(),
                    vc.data_ptr(), o.data_ptr(), lse.data_ptr<float>(),
                    drow.data_ptr<float>(), dq.data_ptr(), dk.data_ptr(),
                    dv.data_ptr(), sq, sk, b, nh, ng, d, (float)scale,
                    causal, cur_stream());
  }
  return {dq, dk, dv};
}

// attn_bwd writing dQ/dK/dV INTO caller-provided (possibly strided)
// buffers — the flash backward can then fill the fused [s, b, g,
// (rep+2)*hn] QKV-grad buffer directly and _SplitQKV.backward skips its
// three slice copies (~1% of a GPT step).  Requires the round-2 kernel
// path; returns false when the shape would fall back to v1 (caller uses
// attn_bwd + copies instead).  Out tensors must have d contiguous.
bool attn_bwd_into(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                   torch::Tensor v, torch::Tensor o, torch::Tensor lse,
                   double scale, bool causal, torch::Tensor dq,
                   torch::Tensor dk, torch::Tensor dv) {
  check_bf16_any(dout, "dout");
  check_bf16_any(q, "q");
  check_bf16_any(k, "k");
  check_bf16_any(v, "v");
  check_bf16_any(dq, "dq");
  check_bf16_any(dk, "dk");
  check_bf16_any(dv, "dv");
  const int sq = (int)q.size(0), b = (int)q.size(1), nh = (int)q.size(2),
            d = (int)q.size(3);
  const int sk = (int)k.size(0), ng = (int)k.size(2);
  long qs[3], ks[3], vs[3], dqs[3], dks[3], dvs[3];
  if (!((d == 128 || d == 64) && sq % 256 == 0 && sk % 256 == 0 &&
        attn_strides(q, qs) && attn_strides(k, ks) && attn_strides(v, vs) &&
        attn_strides(dq, dqs) && attn_strides(dk, dks) &&
        attn_strides(dv, dvs)))
    return false;
  // dq is written with 16-byte vector stores: its strides and base
  // pointer must be 8-element aligned
  if (dqs[0] % 8 || dqs[1] % 8 || dqs[2] % 8 ||
      (reinterpret_cast<uintptr_t>(dq.data_ptr()) & 15))
    return false;
  TORCH_CHECK(dq.sizes() == q.sizes() && dk.sizes() == k.sizes() &&
              dv.sizes() == v.sizes(), "attn_bwd_into: out shape mismatch");
  auto drow = torch::empty({(long)sq * b * nh},
                           q.options().dtype(torch::kFloat32));
  auto doc = dout.contiguous();
  launch_attn_bwd2(doc.data_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
                   o.data_ptr(), lse.data_ptr<float>(),
                   drow.data_ptr<float>(), dq.data_ptr(), dk.data_ptr(),
                   dv.data_ptr(), sq, sk, b, nh, ng, d, (float)scale, causal,
                   qs, ks, vs, dqs, dks, dvs, cur_stream());
  return true;
}

// always-v1 backward, kept for A/B benchmarking
std::vector<torch::Tensor> attn_bwd_v1(torch::Tensor dout, torch::Tensor q,
                                       torch::Tensor k, torch::Tensor v,
                                       torch::Tensor o, torch::Tensor lse,
                                       double scale, bool causal) {
  const int sq = (int)q.size(0), b = (int)q.size(1), nh = (int)q.size(2),
            d = (int)q.size(3);
  const int sk = (int)k.size(0), ng = (int)k.size(2);
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto drow = torch::empty({(long)sq * b * nh},
                           q.options().dtype(torch::kFloat32));
  launch_attn_bwd(dout.data_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
                  o.data_ptr(), lse.data_ptr<float>(), drow.data_ptr<float>(),
                  dq.data_ptr(), dk.data_ptr(), dv.data_ptr(), sq, sk, b, nh,
                  ng, d, (float)scale, causal, cur_stream());
  return {dq, dk, dv};
}

std::vector<torch::Tensor> attn_fwd_ablate(torch::Tensor q, torch::Tensor k,
                                           torch::Tensor v, double scale,
                                           long level) {
  const int sq = (int)q.size(0), b = (int)q.size(1), nh = (int)q.size(2),
            d = (int)q.size(3);
  const int sk = (int)k.size(0), ng = (int)k.size(2);
  auto o = torch::empty_like(q);
  auto lse = torch::empty({b, nh, sq}, q.options().dtype(torch::kFloat32));
  launch_attn_fwd_ablate(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                         o.data_ptr(), lse.data_ptr<float>(), sq, sk, b, nh,
                         ng, d, (float)scale, (int)level, cur_stream());
  return {o, lse};
}

// --------------------------------------------------------------------- adam
// states fp32 OR bf16 (precision-aware optimizer); params/grads fp32
static bool adam_states_bf16(const torch::Tensor& m, const torch::Tensor& v) {
  TORCH_CHECK(m.scalar_type() == v.scalar_type(),
              "adam m/v dtypes must match");
  TORCH_CHECK(m.scalar_type() == torch::kFloat32 ||
              m.scalar_type() == torch::kBFloat16,
              "adam states must be fp32 or bf16");
  return m.scalar_type() == torch::kBFloat16;
}

static void* adam_p16_ptr(const c10::optional<torch::Tensor>& p16,
                          const torch::Tensor& p) {
  if (!p16.has_value()) return nullptr;
  TORCH_CHECK(p16->scalar_type() == torch::kBFloat16 &&
              p16->is_contiguous() && p16->numel() == p.numel(),
              "adam p16 must be a contiguous bf16 mirror of the shard");
  return p16->data_ptr();
}

void adamw_flat(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, double lr, double beta1, double beta2,
                double eps, double wd, long step,
                c10::optional<torch::Tensor> p16) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == torch::kFloat32);
  launch_adamw_flat(p.data_ptr<float>(), g.data_ptr<float>(), m.data_ptr(),
                    v.data_ptr(), adam_states_bf16(m, v),
                    adam_p16_ptr(p16, p), p.numel(),
                    (float)lr, (float)beta1, (float)beta2, (float)eps,
                    (float)wd, (int)step, cur_stream());
}

torch::Tensor selective_scan_fwd(torch::Tensor x, torch::Tensor dt,
                                 torch::Tensor A, torch::Tensor B,
                                 torch::Tensor C, torch::Tensor D,
                                 torch::Tensor h) {
  check_bf16(x, "x");
  check_bf16(dt, "dt");
  check_bf16(B, "B");
  check_bf16(C, "C");
  TORCH_CHECK(A.dtype() == torch::kFloat32 && D.dtype() == torch::kFloat32 &&
              h.dtype() == torch::kFloat32);
  TORCH_CHECK(x.is_contiguous() && dt.is_contiguous() && B.is_contiguous() &&
              C.is_contiguous() && h.is_contiguous());
  const int b = (int)x.size(0), l = (int)x.size(1), d = (int)x.size(2);
  const int n = (int)A.size(1);
  auto y = torch::empty_like(x);
  launch_selective_scan_fwd(x.data_ptr(), dt.data_ptr(),
                            A.data_ptr<float>(), B.data_ptr(), C.data_ptr(),
                            D.data_ptr<float>(), h.data_ptr<float>(),
                            y.data_ptr(), b, l, d, n,
                            at::cuda::getCurrentCUDAStream());
  return y;
}

torch::Tensor ce_rowmax(torch::Tensor logits) {
  check_bf16(logits, "logits");
  TORCH_CHECK(logits.is_contiguous() && logits.dim() == 2);
  const long rows = logits.size(0);
  const int v = (int)logits.size(1);
  TORCH_CHECK(v % 8 == 0);
  auto out = torch::empty({rows}, logits.options().dtype(torch::kFloat32));
  launch_ce_rowmax(logits.data_ptr(), out.data_ptr<float>(), rows, v,
                   cur_stream());
  return out;
}

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor rowmax,
                                  torch::Tensor target) {
  check_bf16(logits, "logits");
  TORCH_CHECK(target.dtype() == torch::kInt32 && target.is_contiguous());
  const long rows = logits.size(0);
  const int v = (int)logits.size(1);
  auto sumexp = torch::empty({rows},
                             logits.options().dtype(torch::kFloat32));
  auto predicted = torch::zeros({rows},
                                logits.options().dtype(torch::kFloat32));
  launch_ce_fwd(logits.data_ptr(), rowmax.data_ptr<float>(),
                target.data_ptr<int>(), sumexp.data_ptr<float>(),
                predicted.data_ptr<float>(), rows, v, cur_stream());
  return {sumexp, predicted};
}

torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor rowmax,
                     torch::Tensor sumexp, torch::Tensor target,
                     torch::Tensor grad_row) {
  check_bf16(logits, "logits");
  const long rows = logits.size(0);
  const int v = (int)logits.size(1);
  auto dlogits = torch::empty_like(logits);
  launch_ce_bwd(logits.data_ptr(), rowmax.data_ptr<float>(),
                sumexp.data_ptr<float>(), target.data_ptr<int>(),
                grad_row.contiguous().data_ptr<float>(), dlogits.data_ptr(),
                rows, v, cur_stream());
  return dlogits;
}

void adamw_flat_ranged(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                       torch::Tensor v, torch::Tensor nw_s, torch::Tensor nw_e,
                       double lr, double beta1, double beta2, double eps,
                       double wd, long step,
                       c10::optional<torch::Tensor> p16) {
  TORCH_CHECK(p.is_cuda() && p.dtype() == torch::kFloat32);
  TORCH_CHECK(nw_s.is_cuda() && nw_s.dtype() == torch::kInt64 &&
              nw_s.is_contiguous());
  TORCH_CHECK(nw_e.sizes() == nw_s.sizes());
  launch_adamw_flat_ranged(
      p.data_ptr<float>(), g.data_ptr<float>(), m.data_ptr(), v.data_ptr(),
      adam_states_bf16(m, v), adam_p16_ptr(p16, p), nw_s.data_ptr<long>(),
      nw_e.data_ptr<long>(),
      (int)nw_s.numel(), p.numel(), (float)lr, (float)beta1, (float)beta2,
      (float)eps, (float)wd, (int)step,
      at::cuda::getCurrentCUDAStream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("rmsnorm_fwd", &rmsnorm_fwd);
  mod.def("rmsnorm_bwd", &rmsnorm_bwd, pybind11::arg("dy"), pybind11::arg("x"),
          pybind11::arg("w"), pybind11::arg("invrms"),
          pybind11::arg("main_grad_w") = pybind11::none(),
          pybind11::arg("dres") = pybind11::none());
  mod.def("layernorm_fwd", &layernorm_fwd);
  mod.def("layernorm_bwd", &layernorm_bwd, pybind11::arg("dy"),
          pybind11::arg("x"), pybind11::arg("w"), pybind11::arg("mean"),
          pybind11::arg("invstd"),
          pybind11::arg("main_grad_w") = pybind11::none(),
          pybind11::arg("main_grad_b") = pybind11::none(),
          pybind11::arg("dres") = pybind11::none());
  mod.def("bias_gelu_fwd", &bias_gelu_fwd);
  mod.def("bias_gelu_bwd", &bias_gelu_bwd);
  mod.def("bias_swiglu_fwd", &bias_swiglu_fwd);
  mod.def("bias_swiglu_bwd", &bias_swiglu_bwd);
  mod.def("bias_geglu_fwd", &bias_geglu_fwd);
  mod.def("bias_geglu_bwd", &bias_geglu_bwd);
  mod.def("rope_fwd", &rope_fwd);
  mod.def("rope_bwd", &rope_bwd);
  mod.def("scaled_upper_triang_masked_softmax_fwd",
          &scaled_upper_triang_masked_softmax_fwd);
  mod.def("scaled_softmax_fwd", &scaled_softmax_fwd);
  mod.def("scaled_masked_softmax_fwd", &scaled_masked_softmax_fwd);
  mod.def("scaled_softmax_bwd", &scaled_softmax_bwd);
  mod.def("adamw_flat", &adamw_flat, pybind11::arg("p"),
          pybind11::arg("g"), pybind11::arg("m"), pybind11::arg("v"),
          pybind11::arg("lr"), pybind11::arg("beta1"),
          pybind11::arg("beta2"), pybind11::arg("eps"),
          pybind11::arg("wd"), pybind11::arg("step"),
          pybind11::arg("p16") = pybind11::none());
  mod.def("selective_scan_fwd", &selective_scan_fwd);
  mod.def("attn_fwd_t", &attn_fwd_t);
  mod.def("attn_fwd2", &attn_fwd2);
  mod.def("quantize_rows_e4m3", &quantize_rows_e4m3);
  mod.def("quantize_transpose_e4m3", &quantize_transpose_e4m3);
  mod.def("bias_dropout_add_fwd", &bias_dropout_add_fwd);
  mod.def("dropout_bwd", &dropout_bwd);
  mod.def("moe_combine_fwd", &moe_combine_fwd);
  mod.def("moe_combine_bwd", &moe_combine_bwd);
  mod.def("attn_bwd_v1", &attn_bwd_v1);
  mod.def("attn_bwd_into", &attn_bwd_into);
  mod.def("ce_rowmax", &ce_rowmax);
  mod.def("gemm_nt", &gemm_nt);
  mod.def("gemm_nn", &gemm_nn);
  mod.def("ce_fwd", &ce_fwd);
  mod.def("ce_bwd", &ce_bwd);
  mod.def("scaled_upper_triang_masked_softmax_bwd",
          &scaled_upper_triang_masked_softmax_bwd);
  mod.def("bias_add_residual", &bias_add_residual);
  mod.def("adamw_flat_ranged", &adamw_flat_ranged, pybind11::arg("p"),
          pybind11::arg("g"), pybind11::arg("m"), pybind11::arg("v"),
          pybind11::arg("nw_s"), pybind11::arg("nw_e"),
          pybind11::arg("lr"), pybind11::arg("beta1"),
          pybind11::arg("beta2"), pybind11::arg("eps"),
          pybind11::arg("wd"), pybind11::arg("step"),
          pybind11::arg("p16") = pybind11::none());
  mod.def("wgrad_accum", &wgrad_accum);
  mod.def("wgrad_accum_bgrad", &wgrad_accum_bgrad);
  mod.def("attn_fwd", &attn_fwd);
  mod.def("attn_fwd_ablate", &attn_fwd_ablate);
  mod.def("attn_bwd", &attn_bwd);
  mod.def("colsum_accum", &colsum_accum);
  mod.def("embedding_bwd_accum", &embedding_bwd_accum);
}
