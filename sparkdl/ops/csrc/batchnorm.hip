// Fused BatchNorm(+residual add)(+ReLU), bf16 NHWC activations with fp32
// statistics and affine parameters (the BN+ReLU
// epilogue of the ResNet conv blocks).
//
// channels_last [N,C,H,W] tensors are exactly [rows=N*H*W, C] row-major,
// so the channel dim is the contiguous innermost axis. All kernels use
// short8 (16 B/lane) vector access; per-channel reductions accumulate in
// registers over grid-strided rows and fold with ONE global atomic per
// thread (Guideline 12).
//
// Replaces the MIOpen BN + eager ReLU + eager residual-add chain (and
// their autocast fp32 round-trips) with:
//   fwd: stats (1 pass over x) -> finalize (C threads) -> apply (1 pass)
//   bwd: reduce (1 pass) -> finalize -> apply (1 pass)
// vs MIOpen's separate BN fwd/bwd + ReLU fwd/bwd + add kernels.

#include "common.hip.h"
#include "kernels.h"

namespace {

constexpr int kBlock = 256;
constexpr int kVec = 8;

// --------------------------------------------------------------------
// Forward
// --------------------------------------------------------------------

// Fold a per-thread [kVec] accumulator pair into the block's LDS tile
// (one slot per column), then ONE global atomic per column per block —
// without this, 2048 blocks x 256 threads of same-address global
// atomics serialize on small-C layers (Guideline 12).
__device__ __forceinline__ void bn_block_fold(
    float* lds_a, float* lds_b, const float* a, const float* b,
    int lane_col, int cols, float* out_a, float* out_b) {
  for (int c = threadIdx.x; c < cols; c += kBlock) {
    lds_a[c] = 0.f;
    lds_b[c] = 0.f;
  }
  __syncthreads();
#pragma unroll
  for (int j = 0; j < kVec; ++j) {
    atomicAdd(&lds_a[lane_col + j], a[j]);  // LDS atomics: per-CU, cheap
    atomicAdd(&lds_b[lane_col + j], b[j]);
  }
  __syncthreads();
  for (int c = threadIdx.x; c < cols; c += kBlock) {
    atomicAdd(&out_a[c], lds_a[c]);
    atomicAdd(&out_b[c], lds_b[c]);
  }
}

// scratch layout: [0..C) sum, [C..2C) sumsq  (pre-zeroed)
__global__ __launch_bounds__(kBlock) void bn_stats_k(
    const short* __restrict__ x, float* __restrict__ scratch,
    long long rows, int cols) {
  __shared__ float lds[2 * 2048];
  const int tpr = cols / kVec;             // threads per row (cols%8==0)
  const int rpb = kBlock / min(tpr, kBlock);
  const int lane_col = (threadIdx.x % tpr) * kVec;
  const int row_off = threadIdx.x / tpr;

  float s[kVec] = {0.f}, ss[kVec] = {0.f};
  if (row_off < rpb) {
    for (long long r = (long long)blockIdx.x * rpb + row_off; r < rows;
         r += (long long)gridDim.x * rpb) {
      const short8 v = *(const short8*)(x + r * cols + lane_col);
#pragma unroll
      for (int j = 0; j < kVec; ++j) {
        const float f = bf2f(v[j]);
        s[j] += f;
        ss[j] += f * f;
      }
    }
  }
  bn_block_fold(lds, lds + cols, s, ss, lane_col, cols, scratch,
                scratch + cols);
}

// One thread per channel: batch stats, running-stat update, and the
// fused apply coefficients scale=gamma*rstd, shift=beta-mean*scale.
__global__ void bn_finalize_k(
    const float* __restrict__ scratch, const float* __restrict__ gamma,
    const float* __restrict__ beta, float* __restrict__ running_mean,
    float* __restrict__ running_var, float* __restrict__ save_mean,
    float* __restrict__ save_rstd, float* __restrict__ scale,
    float* __restrict__ shift, long long rows, int cols, float momentum,
    float eps, int training) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= cols) return;
  float mean, rstd;
  if (training) {
    mean = scratch[c] / rows;
    const float var = fmaxf(scratch[cols + c] / rows - mean * mean, 0.f);
    rstd = rsqrtf(var + eps);
    // torch semantics: running_var tracks the UNBIASED batch variance.
    const float unbiased = rows > 1 ? var * rows / (rows - 1) : var;
    running_mean[c] += momentum * (mean - running_mean[c]);
    running_var[c] += momentum * (unbiased - running_var[c]);
    save_mean[c] = mean;
    save_rstd[c] = rstd;
  } else {
    mean = running_mean[c];
    rstd = rsqrtf(running_var[c] + eps);
    save_mean[c] = mean;
    save_rstd[c] = rstd;
  }
  const float sc = gamma[c] * rstd;
  scale[c] = sc;
  shift[c] = beta[c] - mean * sc;
}

// y = relu?(scale*x + shift [+ residual]); block processes whole rows.
// With RELU, the activation mask is packed 1 bit/elem (one byte per
// vec8 lane slot) so backward never re-reads y.
template <bool RELU, bool RES>
__global__ __launch_bounds__(kBlock) void bn_apply_k(
    const short* __restrict__ x, const short* __restrict__ res,
    const float* __restrict__ scale, const float* __restrict__ shift,
    short* __restrict__ y, unsigned char* __restrict__ mask,
    long long rows, int cols) {
  const int tpr = cols / kVec;
  const int rpb = kBlock / min(tpr, kBlock);
  const int lane_col = (threadIdx.x % tpr) * kVec;
  const int row_off = threadIdx.x / tpr;
  if (row_off >= rpb) return;
  const float4v sc0 = *(const float4v*)(scale + lane_col);
  const float4v sc1 = *(const float4v*)(scale + lane_col + 4);
  const float4v sh0 = *(const float4v*)(shift + lane_col);
  const float4v sh1 = *(const float4v*)(shift + lane_col + 4);

  for (long long r = (long long)blockIdx.x * rpb + row_off; r < rows;
       r += (long long)gridDim.x * rpb) {
    const long long base = r * cols + lane_col;
    const short8 v = *(const short8*)(x + base);
    short8 o;
    unsigned char mbits = 0;
#pragma unroll
    for (int j = 0; j < kVec; ++j) {
      float f = bf2f(v[j]) * (j < 4 ? sc0[j] : sc1[j - 4]) +
                (j < 4 ? sh0[j] : sh1[j - 4]);
      if (RES) f += bf2f(res[base + j]);
      if (RELU) {
        if (f > 0.f) mbits |= (unsigned char)(1u << j);
        f = fmaxf(f, 0.f);
      }
      o[j] = f2bf(f);
    }
    *(short8*)(y + base) = o;
    if (RELU) mask[(r * cols + lane_col) / kVec] = mbits;
  }
}

// --------------------------------------------------------------------
// Backward
// --------------------------------------------------------------------
// dym = dy * relu_mask (mask from saved output y > 0).
// scratch: [0..C) sum(dym), [C..2C) sum(dym * xhat)   (pre-zeroed)

template <bool RELU>
__global__ __launch_bounds__(kBlock) void bn_bwd_reduce_k(
    const short* __restrict__ x, const unsigned char* __restrict__ mask,
    const short* __restrict__ dy, const float* __restrict__ save_mean,
    const float* __restrict__ save_rstd, float* __restrict__ scratch,
    long long rows, int cols) {
  __shared__ float lds[2 * 2048];
  const int tpr = cols / kVec;
  const int rpb = kBlock / min(tpr, kBlock);
  const int lane_col = (threadIdx.x % tpr) * kVec;
  const int row_off = threadIdx.x / tpr;

  float s1[kVec] = {0.f}, s2[kVec] = {0.f};
  if (row_off < rpb) {
    const float4v m0 = *(const float4v*)(save_mean + lane_col);
    const float4v m1 = *(const float4v*)(save_mean + lane_col + 4);
    const float4v r0 = *(const float4v*)(save_rstd + lane_col);
    const float4v r1 = *(const float4v*)(save_rstd + lane_col + 4);
    for (long long r = (long long)blockIdx.x * rpb + row_off; r < rows;
         r += (long long)gridDim.x * rpb) {
      const long long base = r * cols + lane_col;
      const short8 xv = *(const short8*)(x + base);
      const short8 dv = *(const short8*)(dy + base);
      unsigned char mbits = 0xffu;
      if (RELU) mbits = mask[base / kVec];
#pragma unroll
      for (int j = 0; j < kVec; ++j) {
        float d = bf2f(dv[j]);
        if (RELU && !(mbits & (1u << j))) d = 0.f;
        const float mean = j < 4 ? m0[j] : m1[j - 4];
        const float rstd = j < 4 ? r0[j] : r1[j - 4];
        const float xh = (bf2f(xv[j]) - mean) * rstd;
        s1[j] += d;
        s2[j] += d * xh;
      }
    }
  }
  bn_block_fold(lds, lds + cols, s1, s2, lane_col, cols, scratch,
                scratch + cols);
}

// dgamma = sum(dym*xhat), dbeta = sum(dym); coefficients for the apply.
__global__ void bn_bwd_finalize_k(
    const float* __restrict__ scratch, const float* __restrict__ gamma,
    const float* __restrict__ save_rstd, float* __restrict__ dgamma,
    float* __restrict__ dbeta, float* __restrict__ c1,
    float* __restrict__ c2, float* __restrict__ c3, long long rows,
    int cols, int training) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= cols) return;
  const float sum_dym = scratch[c];
  const float sum_dymxh = scratch[cols + c];
  dgamma[c] = sum_dymxh;
  dbeta[c] = sum_dym;
  c1[c] = gamma[c] * save_rstd[c];
  if (training) {
    c2[c] = sum_dym / rows;
    c3[c] = sum_dymxh / rows;
  } else {
    c2[c] = 0.f;  // eval mode: stats are constants
    c3[c] = 0.f;
  }
}

// dx = c1*(dym - c2 - xhat*c3); optional dres = dym.
template <bool RELU, bool RES>
__global__ __launch_bounds__(kBlock) void bn_bwd_apply_k(
    const short* __restrict__ x, const unsigned char* __restrict__ mask,
    const short* __restrict__ dy, const float* __restrict__ save_mean,
    const float* __restrict__ save_rstd, const float* __restrict__ c1,
    const float* __restrict__ c2, const float* __restrict__ c3,
    short* __restrict__ dx, short* __restrict__ dres, long long rows,
    int cols) {
  const int tpr = cols / kVec;
  const int rpb = kBlock / min(tpr, kBlock);
  const int lane_col = (threadIdx.x % tpr) * kVec;
  const int row_off = threadIdx.x / tpr;
  if (row_off >= rpb) return;
  const float4v m0 = *(const float4v*)(save_mean + lane_col);
  const float4v m1 = *(const float4v*)(save_mean + lane_col + 4);
  const float4v r0 = *(const float4v*)(save_rstd + lane_col);
  const float4v r1 = *(const float4v*)(save_rstd + lane_col + 4);
  const float4v a0 = *(const float4v*)(c1 + lane_col);
  const float4v a1 = *(const float4v*)(c1 + lane_col + 4);
  const float4v b0 = *(const float4v*)(c2 + lane_col);
  const float4v b1 = *(const float4v*)(c2 + lane_col + 4);
  const float4v g0 = *(const float4v*)(c3 + lane_col);
  const float4v g1 = *(const float4v*)(c3 + lane_col + 4);

  for (long long r = (long long)blockIdx.x * rpb + row_off; r < rows;
       r += (long long)gridDim.x * rpb) {
    const long long base = r * cols + lane_col;
    const short8 xv = *(const short8*)(x + base);
    const short8 dv = *(const short8*)(dy + base);
    unsigned char mbits = 0xffu;
    if (RELU) mbits = mask[base / kVec];
    short8 odx, odr;
#pragma unroll
    for (int j = 0; j < kVec; ++j) {
      float d = bf2f(dv[j]);
      if (RELU && !(mbits & (1u << j))) d = 0.f;
      const float mean = j < 4 ? m0[j] : m1[j - 4];
      const float rstd = j < 4 ? r0[j] : r1[j - 4];
      const float xh = (bf2f(xv[j]) - mean) * rstd;
      const float k1 = j < 4 ? a0[j] : a1[j - 4];
      const float k2 = j < 4 ? b0[j] : b1[j - 4];
      const float k3 = j < 4 ? g0[j] : g1[j - 4];
      odx[j] = f2bf(k1 * (d - k2 - xh * k3));
      if (RES) odr[j] = f2bf(d);
    }
    *(short8*)(dx + base) = odx;
    if (RES) *(short8*)(dres + base) = odr;
  }
}

int bn_grid(long long rows, int cols) {
  const int tpr = cols / kVec;
  const int rpb = kBlock / min(tpr, kBlock);
  long long blocks = (rows + rpb - 1) / rpb;
  return (int)min(blocks, 2048LL);
}

}  // namespace

void launch_bn_fwd(const short* x, const short* res, const float* gamma,
                   const float* beta, float* running_mean,
                   float* running_var, float* save_mean, float* save_rstd,
                   float* scratch, short* y, unsigned char* mask,
                   long long rows, int cols, float momentum, float eps,
                   bool training, bool relu, hipStream_t stream) {
  const int grid = bn_grid(rows, cols);
  if (training)
    hipLaunchKernelGGL(bn_stats_k, dim3(grid), dim3(kBlock), 0, stream, x,
                       scratch, rows, cols);
  // scale/shift reuse scratch[2C..4C)
  float* scale = scratch + 2 * cols;
  float* shift = scratch + 3 * cols;
  hipLaunchKernelGGL(bn_finalize_k, dim3((cols + 255) / 256), dim3(256), 0,
                     stream, scratch, gamma, beta, running_mean,
                     running_var, save_mean, save_rstd, scale, shift, rows,
                     cols, momentum, eps, (int)training);
  auto ap = relu ? (res ? bn_apply_k<true, true> : bn_apply_k<true, false>)
                 : (res ? bn_apply_k<false, true>
                        : bn_apply_k<false, false>);
  hipLaunchKernelGGL(ap, dim3(grid), dim3(kBlock), 0, stream, x, res,
                     scale, shift, y, mask, rows, cols);
}

void launch_bn_bwd(const short* x, const unsigned char* mask,
                   const short* dy, const float* gamma,
                   const float* save_mean, const float* save_rstd,
                   float* scratch, float* dgamma, float* dbeta, short* dx,
                   short* dres, long long rows, int cols, bool training,
                   bool relu, hipStream_t stream) {
  const int grid = bn_grid(rows, cols);
  auto rk = relu ? bn_bwd_reduce_k<true> : bn_bwd_reduce_k<false>;
  hipLaunchKernelGGL(rk, dim3(grid), dim3(kBlock), 0, stream, x, mask, dy,
                     save_mean, save_rstd, scratch, rows, cols);
  float* c1 = scratch + 2 * cols;
  float* c2 = scratch + 3 * cols;
  float* c3 = scratch + 4 * cols;
  hipLaunchKernelGGL(bn_bwd_finalize_k, dim3((cols + 255) / 256), dim3(256),
                     0, stream, scratch, gamma, save_rstd, dgamma, dbeta,
                     c1, c2, c3, rows, cols, (int)training);
  auto ak = relu ? (dres ? bn_bwd_apply_k<true, true>
                         : bn_bwd_apply_k<true, false>)
                 : (dres ? bn_bwd_apply_k<false, true>
                         : bn_bwd_apply_k<false, false>);
  hipLaunchKernelGGL(ak, dim3(grid), dim3(kBlock), 0, stream, x, mask, dy,
                     save_mean, save_rstd, c1, c2, c3, dx, dres, rows,
                     cols);
}
