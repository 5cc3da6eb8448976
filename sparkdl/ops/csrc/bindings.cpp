// Torch extension bindings for sparkdl's CDNA4 kernels (sparkdl._C).
//
// Host-only translation layer: validates tensors, allocates outputs and
// workspaces through ATen, and launches the pure-HIP kernels declared in
// kernels.h on the current HIP stream.

#include <torch/extension.h>
// ROCm torch masquerades HIP devices as DeviceType::CUDA; the plain
// c10::hip guard/stream APIs reject such devices, so the
// MasqueradingAsCUDA variants are the native entry points.
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <ATen/hip/impl/HIPGuardImplMasqueradingAsCUDA.h>

#include <cmath>
#include <vector>

#include "kernels.h"

namespace {

using DeviceGuard = c10::hip::HIPGuardMasqueradingAsCUDA;

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

#define CHECK_BF16_CUDA(t)                                            \
  TORCH_CHECK((t).is_cuda(), #t " must be on GPU");                   \
  TORCH_CHECK((t).scalar_type() == at::kBFloat16, #t " must be bf16");\
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

#define CHECK_F32_CUDA(t)                                             \
  TORCH_CHECK((t).is_cuda(), #t " must be on GPU");                   \
  TORCH_CHECK((t).scalar_type() == at::kFloat, #t " must be fp32");   \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

const short* bf_ptr(const at::Tensor& t) {
  return reinterpret_cast<const short*>(t.data_ptr<at::BFloat16>());
}
short* bf_ptr_mut(at::Tensor& t) {
  return reinterpret_cast<short*>(t.data_ptr<at::BFloat16>());
}

// ---------------------------------------------------------------------
// Fused optimizers (multi-tensor table built Python-side as blobs)
// ---------------------------------------------------------------------

void fused_adamw_(const at::Tensor& chunks_blob, const at::Tensor& bmap,
                  int64_t nblocks, double lr, double beta1, double beta2,
                  double eps, double weight_decay, at::Tensor step_counter,
                  at::Tensor coeffs) {
  TORCH_CHECK(chunks_blob.is_cuda() && bmap.is_cuda());
  TORCH_CHECK(step_counter.is_cuda() &&
              step_counter.scalar_type() == at::kLong);
  TORCH_CHECK(coeffs.is_cuda() && coeffs.scalar_type() == at::kFloat &&
              coeffs.numel() >= 2);
  DeviceGuard guard(chunks_blob.device());
  launch_fused_adamw(
      reinterpret_cast<const TensorChunk*>(chunks_blob.data_ptr()),
      reinterpret_cast<const int2*>(bmap.data_ptr()), (int)nblocks,
      (float)lr, (float)beta1, (float)beta2, (float)eps,
      (float)weight_decay,
      reinterpret_cast<long long*>(step_counter.data_ptr<int64_t>()),
      coeffs.data_ptr<float>(), cur_stream());
}

void zero_grads_(const at::Tensor& chunks_blob, const at::Tensor& bmap,
                 int64_t nblocks) {
  TORCH_CHECK(chunks_blob.is_cuda() && bmap.is_cuda());
  DeviceGuard guard(chunks_blob.device());
  launch_zero_grads(
      reinterpret_cast<const TensorChunk*>(chunks_blob.data_ptr()),
      reinterpret_cast<const int2*>(bmap.data_ptr()), (int)nblocks,
      cur_stream());
}

void fused_sgd_(const at::Tensor& chunks_blob, const at::Tensor& bmap,
                int64_t nblocks, double lr, double momentum,
                double weight_decay, bool nesterov, bool first_step) {
  TORCH_CHECK(chunks_blob.is_cuda() && bmap.is_cuda());
  DeviceGuard guard(chunks_blob.device());
  launch_fused_sgd(
      reinterpret_cast<const TensorChunk*>(chunks_blob.data_ptr()),
      reinterpret_cast<const int2*>(bmap.data_ptr()), (int)nblocks,
      (float)lr, (float)momentum, (float)weight_decay, nesterov,
      first_step, cur_stream());
}

// ---------------------------------------------------------------------
// LayerNorm
// ---------------------------------------------------------------------

std::vector<at::Tensor> layernorm_fwd(const at::Tensor& x,
                                      const at::Tensor& gamma,
                                      const at::Tensor& beta, double eps) {
  CHECK_BF16_CUDA(x);
  CHECK_F32_CUDA(gamma);
  CHECK_F32_CUDA(beta);
  DeviceGuard guard(x.device());
  const int cols = (int)x.size(-1);
  const int rows = (int)(x.numel() / cols);
  TORCH_CHECK(gamma.numel() == cols && beta.numel() == cols);

  auto y = at::empty_like(x);
  auto f32 = x.options().dtype(at::kFloat);
  auto mean = at::empty({rows}, f32);
  auto rstd = at::empty({rows}, f32);
  launch_layernorm_fwd(bf_ptr(x), gamma.data_ptr<float>(),
                       beta.data_ptr<float>(), bf_ptr_mut(y),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       rows, cols, (float)eps, cur_stream());
  return {y, mean, rstd};
}

std::vector<at::Tensor> layernorm_bwd(const at::Tensor& x,
                                      const at::Tensor& dy,
                                      const at::Tensor& gamma,
                                      const at::Tensor& mean,
                                      const at::Tensor& rstd) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(dy);
  CHECK_F32_CUDA(gamma);
  DeviceGuard guard(x.device());
  const int cols = (int)x.size(-1);
  const int rows = (int)(x.numel() / cols);
  TORCH_CHECK(cols <= 8192, "layernorm_bwd LDS partials support cols<=8192");

  auto dx = at::empty_like(x);
  auto f32 = x.options().dtype(at::kFloat);
  const int blocks = layernorm_bwd_part_rows(rows);
  const int ws_rows = blocks * 4;  // one partial row per wave
  auto dgamma_part = at::empty({ws_rows, cols}, f32);
  auto dbeta_part = at::empty({ws_rows, cols}, f32);
  // zero-init: the 2D reduce kernel folds row-splits with atomics
  auto dgamma = at::zeros({cols}, f32);
  auto dbeta = at::zeros({cols}, f32);
  launch_layernorm_bwd(bf_ptr(x), bf_ptr(dy), gamma.data_ptr<float>(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       bf_ptr_mut(dx), dgamma_part.data_ptr<float>(),
                       dbeta_part.data_ptr<float>(), blocks, rows, cols,
                       cur_stream());
  launch_layernorm_reduce_parts(
      dgamma_part.data_ptr<float>(), dbeta_part.data_ptr<float>(),
      dgamma.data_ptr<float>(), dbeta.data_ptr<float>(), ws_rows, cols,
      cur_stream());
  return {dx, dgamma, dbeta};
}

// ---------------------------------------------------------------------
// BatchNorm (+add) (+ReLU), NHWC bf16
// ---------------------------------------------------------------------

std::vector<at::Tensor> bn_fwd(const at::Tensor& x,
                               const c10::optional<at::Tensor>& res,
                               const at::Tensor& gamma,
                               const at::Tensor& beta,
                               at::Tensor running_mean,
                               at::Tensor running_var, double momentum,
                               double eps, bool training, bool relu) {
  CHECK_BF16_CUDA(x);
  CHECK_F32_CUDA(gamma);
  CHECK_F32_CUDA(beta);
  DeviceGuard guard(x.device());
  const int cols = (int)x.size(-1);
  const long long rows = x.numel() / cols;
  TORCH_CHECK(cols % 8 == 0 && cols <= 2048,
              "bn_fwd requires cols%8==0 and cols<=2048 (NHWC channels)");
  const short* res_ptr = nullptr;
  if (res.has_value()) {
    CHECK_BF16_CUDA(res.value());
    TORCH_CHECK(res->sizes() == x.sizes());
    res_ptr = bf_ptr(res.value());
  }
  auto f32 = x.options().dtype(at::kFloat);
  auto y = at::empty_like(x);
  auto save_mean = at::empty({cols}, f32);
  auto save_rstd = at::empty({cols}, f32);
  auto scratch = at::zeros({5 * cols}, f32);
  at::Tensor mask;
  unsigned char* mask_ptr = nullptr;
  if (relu) {
    // 1 bit per element: backward uses this instead of re-reading y
    mask = at::empty({rows * cols / 8}, x.options().dtype(at::kByte));
    mask_ptr = mask.data_ptr<unsigned char>();
  } else {
    mask = at::empty({0}, x.options().dtype(at::kByte));
  }
  launch_bn_fwd(bf_ptr(x), res_ptr, gamma.data_ptr<float>(),
                beta.data_ptr<float>(), running_mean.data_ptr<float>(),
                running_var.data_ptr<float>(), save_mean.data_ptr<float>(),
                save_rstd.data_ptr<float>(), scratch.data_ptr<float>(),
                bf_ptr_mut(y), mask_ptr, rows, cols, (float)momentum,
                (float)eps, training, relu, cur_stream());
  return {y, save_mean, save_rstd, mask};
}

std::vector<at::Tensor> bn_bwd(const at::Tensor& x,
                               const at::Tensor& mask,
                               const at::Tensor& dy,
                               const at::Tensor& gamma,
                               const at::Tensor& save_mean,
                               const at::Tensor& save_rstd, bool training,
                               bool relu, bool needs_dres) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(dy);
  CHECK_F32_CUDA(gamma);
  DeviceGuard guard(x.device());
  const int cols = (int)x.size(-1);
  const long long rows = x.numel() / cols;
  TORCH_CHECK(!relu || mask.numel() == rows * cols / 8,
              "relu backward needs the forward's activation mask");
  auto f32 = x.options().dtype(at::kFloat);
  auto dx = at::empty_like(x);
  auto dgamma = at::empty({cols}, f32);
  auto dbeta = at::empty({cols}, f32);
  auto scratch = at::zeros({5 * cols}, f32);
  at::Tensor dres;
  short* dres_ptr = nullptr;
  if (needs_dres) {
    dres = at::empty_like(x);
    dres_ptr = bf_ptr_mut(dres);
  }
  launch_bn_bwd(bf_ptr(x),
                relu ? mask.data_ptr<unsigned char>() : nullptr,
                bf_ptr(dy), gamma.data_ptr<float>(),
                save_mean.data_ptr<float>(), save_rstd.data_ptr<float>(),
                scratch.data_ptr<float>(), dgamma.data_ptr<float>(),
                dbeta.data_ptr<float>(), bf_ptr_mut(dx), dres_ptr, rows,
                cols, training, relu, cur_stream());
  if (needs_dres) return {dx, dgamma, dbeta, dres};
  return {dx, dgamma, dbeta};
}

// ---------------------------------------------------------------------
// Bias + GELU
// ---------------------------------------------------------------------

at::Tensor bias_gelu_fwd(const at::Tensor& x, const at::Tensor& bias) {
  CHECK_BF16_CUDA(x);
  CHECK_F32_CUDA(bias);
  DeviceGuard guard(x.device());
  const int cols = (int)x.size(-1);
  const long long rows = x.numel() / cols;
  TORCH_CHECK(bias.numel() == cols);
  auto y = at::empty_like(x);
  launch_bias_gelu_fwd(bf_ptr(x), bias.data_ptr<float>(), bf_ptr_mut(y),
                       rows, cols, cur_stream());
  return y;
}

std::vector<at::Tensor> bias_gelu_bwd(const at::Tensor& x,
                                      const at::Tensor& bias,
                                      const at::Tensor& dy) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(dy);
  CHECK_F32_CUDA(bias);
  DeviceGuard guard(x.device());
  const int cols = (int)x.size(-1);
  const long long rows = x.numel() / cols;
  TORCH_CHECK(cols <= 16384, "bias_gelu_bwd LDS partials support cols<=16384");
  auto dx = at::empty_like(x);
  auto dbias = at::zeros({cols}, x.options().dtype(at::kFloat));
  launch_bias_gelu_bwd(bf_ptr(x), bias.data_ptr<float>(), bf_ptr(dy),
                       bf_ptr_mut(dx), dbias.data_ptr<float>(), rows, cols,
                       cur_stream());
  return {dx, dbias};
}

// ---------------------------------------------------------------------
// MFMA GEMM + bias(+GELU)
// ---------------------------------------------------------------------

std::vector<at::Tensor> gemm_bias_act(const at::Tensor& A,
                                      const at::Tensor& W,
                                      const c10::optional<at::Tensor>& bias,
                                      int64_t act, bool save_z) {
  CHECK_BF16_CUDA(A);
  CHECK_BF16_CUDA(W);
  TORCH_CHECK(A.dim() == 2 && W.dim() == 2 && A.size(1) == W.size(1));
  const int M = (int)A.size(0), K = (int)A.size(1), N = (int)W.size(0);
  TORCH_CHECK(gemm_bias_act_supported(M, N, K),
              "gemm_bias_act requires K % 64 == 0");
  const float* bias_ptr = nullptr;
  if (bias.has_value()) {
    CHECK_F32_CUDA(bias.value());
    TORCH_CHECK(bias->numel() == N);
    bias_ptr = bias->data_ptr<float>();
  }
  DeviceGuard guard(A.device());
  auto C = at::empty({M, N}, A.options());
  at::Tensor Z;
  short* z_ptr = nullptr;
  if (save_z && act != 0) {
    Z = at::empty({M, N}, A.options());
    z_ptr = bf_ptr_mut(Z);
  }
  launch_gemm_bias_act(bf_ptr(A), bf_ptr(W), bias_ptr, bf_ptr_mut(C),
                       z_ptr, M, N, K, (int)act, cur_stream());
  if (z_ptr) return {C, Z};
  return {C};
}

at::Tensor gemm_stream(const at::Tensor& A, const at::Tensor& W) {
  CHECK_BF16_CUDA(A);
  CHECK_BF16_CUDA(W);
  TORCH_CHECK(A.dim() == 2 && W.dim() == 2 && A.size(1) == W.size(1));
  const long long M = A.size(0);
  const int K = (int)A.size(1), N = (int)W.size(0);
  TORCH_CHECK(gemm_stream_supported(M, N, K),
              "gemm_stream requires K in {64,128,256} and N % 64 == 0");
  DeviceGuard guard(A.device());
  auto C = at::empty({M, N}, A.options());
  launch_gemm_stream(bf_ptr(A), bf_ptr(W), bf_ptr_mut(C), M, N, K,
                     cur_stream());
  return C;
}

// ---------------------------------------------------------------------
// Flash attention (D=64)
// ---------------------------------------------------------------------

static const long long* seed_ptr(const c10::optional<at::Tensor>& seed) {
  if (!seed.has_value()) return nullptr;
  TORCH_CHECK(seed->is_cuda() && seed->scalar_type() == at::kLong &&
              seed->numel() >= 1);
  return reinterpret_cast<const long long*>(seed->data_ptr<int64_t>());
}

std::vector<at::Tensor> attn_fwd(const at::Tensor& Q, const at::Tensor& K,
                                 const at::Tensor& V, double dropout_p,
                                 const c10::optional<at::Tensor>& seed) {
  CHECK_BF16_CUDA(Q);
  CHECK_BF16_CUDA(K);
  CHECK_BF16_CUDA(V);
  TORCH_CHECK(Q.dim() == 3 && Q.size(2) == 64 && Q.sizes() == K.sizes() &&
                  Q.sizes() == V.sizes(),
              "attn_fwd expects [BH, S, 64] bf16 q/k/v");
  const int BH = (int)Q.size(0), S = (int)Q.size(1);
  TORCH_CHECK(S % 64 == 0, "attn_fwd requires S % 64 == 0");
  TORCH_CHECK(dropout_p == 0.0 || seed.has_value(),
              "dropout needs a seed tensor");
  DeviceGuard guard(Q.device());
  auto O = at::empty_like(Q);
  auto LSE = at::empty({BH, S}, Q.options().dtype(at::kFloat));
  auto VT = at::empty({BH, 64, S}, Q.options());
  launch_attn_pretranspose(bf_ptr(V), bf_ptr_mut(VT), BH, S, 1, 64,
                           cur_stream());
  launch_attn_fwd(bf_ptr(Q), bf_ptr(K), bf_ptr(VT), bf_ptr_mut(O),
                  LSE.data_ptr<float>(), seed_ptr(seed), BH, S, 1, 64,
                  64, (float)dropout_p, cur_stream());
  return {O, LSE};
}

std::vector<at::Tensor> attn_fwd_packed(
    const at::Tensor& qkv, double dropout_p,
    const c10::optional<at::Tensor>& seed) {
  CHECK_BF16_CUDA(qkv);
  TORCH_CHECK(qkv.dim() == 5 && qkv.size(2) == 3 && qkv.size(4) == 64 &&
                  qkv.is_contiguous(),
              "attn_fwd_packed expects contiguous [B,S,3,H,64] bf16");
  const int B = (int)qkv.size(0), S = (int)qkv.size(1),
            H = (int)qkv.size(3);
  TORCH_CHECK(S % 64 == 0, "attn requires S % 64 == 0");
  TORCH_CHECK(dropout_p == 0.0 || seed.has_value());
  DeviceGuard guard(qkv.device());
  auto O = at::empty({B, S, H * 64}, qkv.options());
  auto LSE = at::empty({B * H, S}, qkv.options().dtype(at::kFloat));
  const short* base = bf_ptr(qkv);
  auto VT = at::empty({B * H, 64, S}, qkv.options());
  launch_attn_pretranspose(base + (long long)2 * H * 64, bf_ptr_mut(VT),
                           B * H, S, H, 3 * H * 64, cur_stream());
  launch_attn_fwd(base, base + (long long)H * 64, bf_ptr(VT),
                  bf_ptr_mut(O), LSE.data_ptr<float>(), seed_ptr(seed),
                  B * H, S, H, 3 * H * 64, H * 64, (float)dropout_p,
                  cur_stream());
  return {O, LSE};
}

std::vector<at::Tensor> attn_bwd(const at::Tensor& Q, const at::Tensor& K,
                                 const at::Tensor& V, const at::Tensor& dO,
                                 const at::Tensor& LSE,
                                 const at::Tensor& Drow, double dropout_p,
                                 const c10::optional<at::Tensor>& seed) {
  CHECK_BF16_CUDA(Q);
  CHECK_BF16_CUDA(dO);
  CHECK_F32_CUDA(LSE);
  CHECK_F32_CUDA(Drow);
  const int BH = (int)Q.size(0), S = (int)Q.size(1);
  DeviceGuard guard(Q.device());
  auto dQ = at::empty_like(Q);
  auto dK = at::empty_like(K);
  auto dV = at::empty_like(V);
  auto T3 = at::empty({3, BH, 64, S}, Q.options());
  short* qt = bf_ptr_mut(T3);
  short* dot = qt + (long long)BH * 64 * S;
  short* ktt = dot + (long long)BH * 64 * S;
  launch_attn_pretranspose(bf_ptr(Q), qt, BH, S, 1, 64, cur_stream());
  launch_attn_pretranspose(bf_ptr(dO), dot, BH, S, 1, 64, cur_stream());
  launch_attn_pretranspose(bf_ptr(K), ktt, BH, S, 1, 64, cur_stream());
  launch_attn_bwd(bf_ptr(Q), bf_ptr(K), bf_ptr(V), bf_ptr(dO), qt, dot,
                  ktt, LSE.data_ptr<float>(), Drow.data_ptr<float>(),
                  bf_ptr_mut(dQ), bf_ptr_mut(dK), bf_ptr_mut(dV),
                  seed_ptr(seed), BH, S, 1, 64, 64, 64, (float)dropout_p,
                  cur_stream());
  return {dQ, dK, dV};
}

at::Tensor attn_bwd_packed(const at::Tensor& qkv, const at::Tensor& dO,
                           const at::Tensor& LSE, const at::Tensor& Drow,
                           double dropout_p,
                           const c10::optional<at::Tensor>& seed) {
  CHECK_BF16_CUDA(qkv);
  CHECK_BF16_CUDA(dO);
  CHECK_F32_CUDA(LSE);
  CHECK_F32_CUDA(Drow);
  TORCH_CHECK(qkv.dim() == 5 && qkv.is_contiguous() &&
              dO.is_contiguous());
  const int B = (int)qkv.size(0), S = (int)qkv.size(1),
            H = (int)qkv.size(3);
  DeviceGuard guard(qkv.device());
  auto dqkv = at::empty_like(qkv);
  const short* base = bf_ptr(qkv);
  short* gbase = bf_ptr_mut(dqkv);
  const int BH = B * H;
  auto T3 = at::empty({3, BH, 64, S}, qkv.options());
  short* qt = bf_ptr_mut(T3);
  short* dot = qt + (long long)BH * 64 * S;
  short* ktt = dot + (long long)BH * 64 * S;
  launch_attn_pretranspose(base, qt, BH, S, H, 3 * H * 64, cur_stream());
  launch_attn_pretranspose(bf_ptr(dO), dot, BH, S, H, H * 64,
                           cur_stream());
  launch_attn_pretranspose(base + (long long)H * 64, ktt, BH, S, H,
                           3 * H * 64, cur_stream());
  launch_attn_bwd(base, base + (long long)H * 64,
                  base + (long long)2 * H * 64, bf_ptr(dO), qt, dot, ktt,
                  LSE.data_ptr<float>(), Drow.data_ptr<float>(), gbase,
                  gbase + (long long)H * 64,
                  gbase + (long long)2 * H * 64, seed_ptr(seed), BH, S,
                  H, 3 * H * 64, H * 64, 3 * H * 64, (float)dropout_p,
                  cur_stream());
  return dqkv;
}

at::Tensor wgrad_splitk(const at::Tensor& dZ, const at::Tensor& X) {
  CHECK_BF16_CUDA(dZ);
  CHECK_BF16_CUDA(X);
  TORCH_CHECK(dZ.dim() == 2 && X.dim() == 2 && dZ.size(0) == X.size(0));
  const long long M = dZ.size(0);
  const int N = (int)dZ.size(1), K = (int)X.size(1);
  TORCH_CHECK(wgrad_splitk_supported(M, N, K),
              "wgrad_splitk requires N%256==0, K%128==0, M%64==0");
  DeviceGuard guard(dZ.device());
  auto dW = at::empty({N, K}, dZ.options().dtype(at::kFloat));
  launch_wgrad_splitk(bf_ptr(dZ), bf_ptr(X), dW.data_ptr<float>(), M, N,
                      K, cur_stream());
  return dW;
}

at::Tensor transpose_bf16(const at::Tensor& X) {
  CHECK_BF16_CUDA(X);
  TORCH_CHECK(X.dim() == 2 && X.is_contiguous());
  DeviceGuard guard(X.device());
  const int R = (int)X.size(0), C = (int)X.size(1);
  auto Y = at::empty({C, R}, X.options());
  launch_transpose_bf16(bf_ptr(X), bf_ptr_mut(Y), R, C, cur_stream());
  return Y;
}

// ---------------------------------------------------------------------
// GBT histogram
// ---------------------------------------------------------------------

at::Tensor gbt_histogram(const at::Tensor& B, const at::Tensor& g,
                         const at::Tensor& h, const at::Tensor& row_list,
                         const at::Tensor& bmap, int64_t n_nodes) {
  TORCH_CHECK(B.is_cuda() && B.scalar_type() == at::kByte &&
              B.is_contiguous());
  CHECK_F32_CUDA(g);
  CHECK_F32_CUDA(h);
  TORCH_CHECK(row_list.is_cuda() && row_list.scalar_type() == at::kInt);
  TORCH_CHECK(bmap.is_cuda() && bmap.scalar_type() == at::kInt &&
              bmap.size(-1) == 4);
  DeviceGuard guard(B.device());
  const int F = (int)B.size(1);
  TORCH_CHECK(F % 8 == 0, "feature dim must be padded to a multiple of 8");
  auto hist = at::zeros({n_nodes, F, 256, 2},
                        B.options().dtype(at::kFloat));
  launch_gbt_histogram(
      B.data_ptr<unsigned char>(), g.data_ptr<float>(),
      h.data_ptr<float>(), row_list.data_ptr<int>(),
      reinterpret_cast<const int4*>(bmap.data_ptr<int>()),
      (int)(bmap.numel() / 4), F, hist.data_ptr<float>(), cur_stream());
  return hist;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "sparkdl MI355X-native kernels (gfx950)";
  m.def("fused_adamw_", &fused_adamw_, "Fused multi-tensor AdamW step");
  m.def("fused_sgd_", &fused_sgd_, "Fused multi-tensor SGD step");
  m.def("zero_grads_", &zero_grads_, "Zero all grads in a chunk table");
  m.def("gemm_bias_act", &gemm_bias_act, "MFMA GEMM + bias(+GELU) (N6)");
  m.def("transpose_bf16", &transpose_bf16, "bf16 2D transpose (dgrad W^T)");
  m.def("gemm_stream", &gemm_stream,
        "Streaming tall-skinny GEMM (1x1 convs)");
  m.def("wgrad_splitk", &wgrad_splitk,
        "Split-M wgrad GEMM (tr_b16 fragment path)");
  m.def("attn_fwd", &attn_fwd, "Flash attention fwd (D=64, bf16)");
  m.def("attn_bwd", &attn_bwd, "Flash attention bwd (dQ/dK/dV)");
  m.def("attn_fwd_packed", &attn_fwd_packed,
        "Flash attention fwd over the packed qkv buffer");
  m.def("attn_bwd_packed", &attn_bwd_packed,
        "Flash attention bwd -> packed dqkv");
  m.def("gbt_histogram", &gbt_histogram, "GBT g/h histogram (N7)");
  m.def("bn_fwd", &bn_fwd, "Fused BatchNorm(+add)(+ReLU) fwd (bf16 NHWC)");
  m.def("bn_bwd", &bn_bwd, "Fused BatchNorm(+add)(+ReLU) bwd (bf16 NHWC)");
  m.def("layernorm_fwd", &layernorm_fwd, "LayerNorm forward (bf16)");
  m.def("layernorm_bwd", &layernorm_bwd, "LayerNorm backward (bf16)");
  m.def("bias_gelu_fwd", &bias_gelu_fwd, "Fused bias+GELU forward (bf16)");
  m.def("bias_gelu_bwd", &bias_gelu_bwd, "Fused bias+GELU backward (bf16)");
  m.attr("_chunk_elems") = kOptChunk;
}
