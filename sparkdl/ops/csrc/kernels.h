// Host-side launch declarations for sparkdl's CDNA4 kernels.
// Implementations in the sibling .hip files; called from bindings.cpp.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

// Multi-tensor chunk descriptor (fused optimizers).
// Mixed-precision regime: when pl != nullptr the trainable parameter is
// bf16 (pl), p is its fp32 MASTER copy, and g points to bf16 gradients
// (reinterpret); the update runs in fp32 and writes both p and pl.
struct TensorChunk {
  float* p;        // parameter (fp32, or fp32 master when pl set)
  const float* g;  // gradient (fp32; bf16 when pl set)
  float* m;        // exp_avg / momentum
  float* v;        // exp_avg_sq (Adam only)
  long long n;     // numel of this tensor
  short* pl;       // bf16 parameter (nullptr in pure-fp32 mode)
};

// Per-block work assignment: x = tensor index, y = chunk start element.
// Built on the host, uploaded once per table rebuild.
constexpr int kOptChunk = 16384;

// step_counter: device int64[1], incremented on-device (hipGraph-safe);
// coeffs: device float[2] scratch receiving {1/bc1, 1/sqrt(bc2)}.
void launch_fused_adamw(const TensorChunk* chunks, const int2* bmap,
                        int nblocks, float lr, float beta1, float beta2,
                        float eps, float weight_decay,
                        long long* step_counter, float* coeffs,
                        hipStream_t stream);

void launch_fused_sgd(const TensorChunk* chunks, const int2* bmap,
                      int nblocks, float lr, float momentum,
                      float weight_decay, bool nesterov, bool first_step,
                      hipStream_t stream);

// LayerNorm over the last dimension, bf16 in/out, fp32 gamma/beta and
// statistics.
void launch_layernorm_fwd(const short* x, const float* gamma,
                          const float* beta, short* y, float* save_mean,
                          float* save_rstd, int rows, int cols, float eps,
                          hipStream_t stream);

void launch_layernorm_bwd(const short* x, const short* dy,
                          const float* gamma, const float* save_mean,
                          const float* save_rstd, short* dx,
                          float* dgamma_part, float* dbeta_part,
                          int part_rows, int rows, int cols,
                          hipStream_t stream);

int layernorm_bwd_part_rows(int rows);

void launch_layernorm_reduce_parts(const float* dgamma_part,
                                   const float* dbeta_part, float* dgamma,
                                   float* dbeta, int part_rows, int cols,
                                   hipStream_t stream);

// Fused bias + GELU (erf form), bf16 activations, fp32 bias
// (SURVEY.md §2.2 N6 epilogue).
void launch_bias_gelu_fwd(const short* x, const float* bias, short* y,
                          long long rows, int cols, hipStream_t stream);

void launch_bias_gelu_bwd(const short* x, const float* bias,
                          const short* dy, short* dx, float* dbias,
                          long long rows, int cols, hipStream_t stream);

// Fused BatchNorm(+residual)(+ReLU), bf16 NHWC, fp32 stats
// (SURVEY.md §2.2 N4/N5). scratch: fp32[5*cols], pre-zeroed [0..2C).
// With relu, the forward packs the activation mask 1 bit/elem
// (uint8[rows*cols/8]) so backward never re-reads y.
void launch_bn_fwd(const short* x, const short* res, const float* gamma,
                   const float* beta, float* running_mean,
                   float* running_var, float* save_mean, float* save_rstd,
                   float* scratch, short* y, unsigned char* mask,
                   long long rows, int cols, float momentum, float eps,
                   bool training, bool relu, hipStream_t stream);

void launch_bn_bwd(const short* x, const unsigned char* mask,
                   const short* dy, const float* gamma,
                   const float* save_mean, const float* save_rstd,
                   float* scratch, float* dgamma, float* dbeta, short* dx,
                   short* dres, long long rows, int cols, bool training,
                   bool relu, hipStream_t stream);

// Hand-written MFMA bf16 GEMM with fused bias(+GELU) epilogue
// (SURVEY.md §2.2 N6): C = act(A[M,K] @ W[N,K]^T + bias), bf16 in/out,
// fp32 accumulation; optional pre-activation save for backward.
void launch_gemm_bias_act(const short* A, const short* W,
                          const float* bias, short* C, short* Z, int M,
                          int N, int K, int act, hipStream_t stream);
bool gemm_bias_act_supported(int M, int N, int K);

// Streaming tall-skinny GEMM (ResNet 1x1 convs): C = A @ W^T, W panel
// resident in LDS, A fragments direct from global. K in {64,128,256},
// N % 64 == 0.
void launch_gemm_stream(const short* A, const short* W, short* C,
                        long long M, int N, int K, hipStream_t stream);
bool gemm_stream_supported(long long M, int N, int K);

// Split-M wgrad GEMM (dW = dZ^T @ X, M-major operands): blocked-LDS
// tr_b16 fragment path, fp32 output with split atomics. Ships behind
// SPARKDL_FUSED_WGRAD=1 (Tensile still measures faster).
void launch_wgrad_splitk(const short* dZ, const short* X, float* dW,
                         long long M, int N, int K, hipStream_t stream);
bool wgrad_splitk_supported(long long M, int N, int K);

// bf16 matrix transpose (dgrad's W^T operand): Y[C,R] = X[R,C]^T.
void launch_transpose_bf16(const short* X, short* Y, int R, int C,
                           hipStream_t stream);

// Flash attention, D=64 heads (SURVEY.md §2.2 N6 attention GEMMs):
// O = softmax(QK^T/sqrt(64))V over [BH, S, 64] bf16, S % 64 == 0.
// LSE ([BH,S] fp32) feeds the flash backward; dropout is a counter-
// hash RNG keyed on a device-resident seed (hipGraph-capture safe).
// rs/ors/grs: element strides between consecutive sequence rows of
// the q/k/v inputs, the O/dO buffer, and the gradient outputs — 64 for
// plain [BH,S,64] tensors, 3*H*64 / H*64 when reading the packed
// [B,S,3,H,64] qkv buffer and writing O as [B,S,H*64] (zero layout
// copies around the attention in the BERT block).
// Pre-transposed operand scratch T[bh][d][s] (built once per call;
// makes every in-kernel staging a linear global_load_lds).
void launch_attn_pretranspose(const short* X, short* T, int BH, int S,
                              int H, int rs, hipStream_t stream);
void launch_attn_fwd(const short* Q, const short* K, const short* VT,
                     short* O, float* LSE, const long long* seed, int BH,
                     int S, int H, int rs, int ors, float dropout_p,
                     hipStream_t stream);
void launch_attn_bwd(const short* Q, const short* K, const short* V,
                     const short* dO, const short* QT, const short* DOT,
                     const short* KT, const float* LSE, const float* Drow,
                     short* dQ, short* dK, short* dV,
                     const long long* seed, int BH, int S, int H, int rs,
                     int ors, int grs, float dropout_p,
                     hipStream_t stream);

// GBT per-(node,feature,bin) gradient/hessian histograms
// (SURVEY.md §2.2 N7). bmap: per-block {node, f0, start, count} over a
// node-sorted row_list; hist: fp32 [n_nodes, F, 256, 2], pre-zeroed.
void launch_gbt_histogram(const unsigned char* B, const float* g,
                          const float* h, const int* row_list,
                          const int4* bmap, int nblocks, int F,
                          float* hist, hipStream_t stream);

// Zero every gradient in a multi-tensor chunk table in one launch
// (replaces the per-tensor zero_grad fill storm).
void launch_zero_grads(const TensorChunk* chunks, const int2* bmap,
                       int nblocks, hipStream_t stream);
