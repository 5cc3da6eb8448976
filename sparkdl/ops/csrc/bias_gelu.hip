// Fused bias + GELU (exact erf form) forward/backward, bf16 activations,
// fp32 bias (SURVEY.md §2.2 N6 epilogue fusion).
//
// Memory-bound elementwise op: fusing the bias add into the GELU pass
// saves one full read+write of the activation tensor vs separate
// (add, gelu) kernels. short8 (16 B/lane) vector path with scalar tail;
// backward accumulates dbias per-block in dynamic LDS, one global atomic
// per column per block (Guideline 12: reduce first, atomics last).

#include "common.hip.h"
#include "kernels.h"

namespace {

constexpr int kBlock = 256;
constexpr float kRsqrt2 = 0.70710678118654752f;
constexpr float kRsqrt2Pi = 0.39894228040143268f;

__device__ __forceinline__ float gelu_f(float z) {
  return 0.5f * z * (1.f + erff(z * kRsqrt2));
}

__device__ __forceinline__ float gelu_grad_f(float z) {
  return 0.5f * (1.f + erff(z * kRsqrt2)) +
         z * kRsqrt2Pi * __expf(-0.5f * z * z);
}

// Row-structured: the column index is tid-derived (no per-element
// 64-bit modulo, which cost ~3x off the HBM roofline in the flat
// grid-stride form).
__global__ __launch_bounds__(kBlock) void bias_gelu_fwd_k(
    const short* __restrict__ x, const float* __restrict__ bias,
    short* __restrict__ y, long long rows, int cols) {
  const int tpr = min((long long)cols / 8, (long long)kBlock);
  const int rpb = kBlock / tpr;
  const int lane_col = (threadIdx.x % tpr) * 8;
  const int row_off = threadIdx.x / tpr;
  if (row_off >= rpb) return;
  for (long long r = (long long)blockIdx.x * rpb + row_off; r < rows;
       r += (long long)gridDim.x * rpb) {
    const short* xr = x + r * cols;
    short* yr = y + r * cols;
    for (int c = lane_col; c < cols; c += tpr * 8) {
      const short8 v = *(const short8*)(xr + c);
      const float4v b0 = *(const float4v*)(bias + c);
      const float4v b1 = *(const float4v*)(bias + c + 4);
      short8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = f2bf(gelu_f(bf2f(v[j]) + (j < 4 ? b0[j] : b1[j - 4])));
      *(short8*)(yr + c) = o;
    }
  }
}

__global__ __launch_bounds__(kBlock) void bias_gelu_fwd_scalar_k(
    const short* __restrict__ x, const float* __restrict__ bias,
    short* __restrict__ y, long long total, int cols) {
  for (long long i = (long long)blockIdx.x * kBlock + threadIdx.x;
       i < total; i += (long long)gridDim.x * kBlock)
    y[i] = f2bf(gelu_f(bf2f(x[i]) + bias[(int)(i % cols)]));
}

extern __shared__ float bg_lds[];  // dbias partials [cols]

// Vector path (cols % 8 == 0): each block processes whole rows; thread t
// owns columns {t*8..t*8+7} + k*kBlock*8 — column ownership is exclusive
// within the block, so the LDS dbias accumulation needs NO atomics.
__global__ __launch_bounds__(kBlock) void bias_gelu_bwd_k(
    const short* __restrict__ x, const float* __restrict__ bias,
    const short* __restrict__ dy, short* __restrict__ dx,
    float* __restrict__ dbias, long long rows, int cols) {
  for (int c = threadIdx.x; c < cols; c += kBlock) bg_lds[c] = 0.f;
  __syncthreads();

  for (long long row = blockIdx.x; row < rows; row += gridDim.x) {
    const short* xr = x + row * cols;
    const short* dyr = dy + row * cols;
    short* dxr = dx + row * cols;
    for (int c = threadIdx.x * 8; c < cols; c += kBlock * 8) {
      const short8 xv = *(const short8*)(xr + c);
      const short8 dv = *(const short8*)(dyr + c);
      const float4v b0 = *(const float4v*)(bias + c);
      const float4v b1 = *(const float4v*)(bias + c + 4);
      short8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float z = bf2f(xv[j]) + (j < 4 ? b0[j] : b1[j - 4]);
        const float g = bf2f(dv[j]) * gelu_grad_f(z);
        o[j] = f2bf(g);
        bg_lds[c + j] += g;  // exclusive column: plain LDS add
      }
      *(short8*)(dxr + c) = o;
    }
  }
  __syncthreads();
  for (int c = threadIdx.x; c < cols; c += kBlock)
    atomicAdd(&dbias[c], bg_lds[c]);
}

__global__ __launch_bounds__(kBlock) void bias_gelu_bwd_scalar_k(
    const short* __restrict__ x, const float* __restrict__ bias,
    const short* __restrict__ dy, short* __restrict__ dx,
    float* __restrict__ dbias, long long rows, int cols) {
  for (int c = threadIdx.x; c < cols; c += kBlock) bg_lds[c] = 0.f;
  __syncthreads();
  for (long long row = blockIdx.x; row < rows; row += gridDim.x) {
    const short* xr = x + row * cols;
    const short* dyr = dy + row * cols;
    short* dxr = dx + row * cols;
    for (int c = threadIdx.x; c < cols; c += kBlock) {
      const float z = bf2f(xr[c]) + bias[c];
      const float g = bf2f(dyr[c]) * gelu_grad_f(z);
      dxr[c] = f2bf(g);
      atomicAdd(&bg_lds[c], g);
    }
  }
  __syncthreads();
  for (int c = threadIdx.x; c < cols; c += kBlock)
    atomicAdd(&dbias[c], bg_lds[c]);
}

}  // namespace

void launch_bias_gelu_fwd(const short* x, const float* bias, short* y,
                          long long rows, int cols, hipStream_t stream) {
  const long long total = rows * cols;
  if (cols % 8 == 0) {
    const int tpr = (int)min((long long)cols / 8, (long long)kBlock);
    const int rpb = kBlock / tpr;
    const int grid =
        (int)min((rows + rpb - 1) / (long long)rpb, 2048LL);
    hipLaunchKernelGGL(bias_gelu_fwd_k, dim3(max(grid, 1)), dim3(kBlock),
                       0, stream, x, bias, y, rows, cols);
  } else {
    const int grid = (int)min((total + kBlock - 1) / kBlock, 2048LL);
    hipLaunchKernelGGL(bias_gelu_fwd_scalar_k, dim3(max(grid, 1)),
                       dim3(kBlock), 0, stream, x, bias, y, total, cols);
  }
}

void launch_bias_gelu_bwd(const short* x, const float* bias,
                          const short* dy, short* dx, float* dbias,
                          long long rows, int cols, hipStream_t stream) {
  const int grid = (int)min(rows, 2048LL);
  const size_t lds = (size_t)cols * sizeof(float);
  if (cols % 8 == 0) {
    hipLaunchKernelGGL(bias_gelu_bwd_k, dim3(max(grid, 1)), dim3(kBlock),
                       lds, stream, x, bias, dy, dx, dbias, rows, cols);
  } else {
    hipLaunchKernelGGL(bias_gelu_bwd_scalar_k, dim3(max(grid, 1)),
                       dim3(kBlock), lds, stream, x, bias, dy, dx, dbias,
                       rows, cols);
  }
}
