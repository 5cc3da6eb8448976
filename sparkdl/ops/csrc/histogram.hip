// GBT (feature, bin) gradient/hessian histogram kernel
// (SURVEY.md §2.2 N7 — the xgboost gpu_hist analogue).
//
// Rows are pre-partitioned by tree node (row_list sorted by node); each
// workgroup owns (node, 8-feature chunk, row slice) from a host-built
// work map. Bins privatize in LDS ([8 features][256 bins][g,h] fp32 =
// 16 KiB) — Guideline 12: block-local accumulation first, one global
// atomic per (feature, bin) at the end. The 8 bin indices per row load
// as a single aligned uint64 (feature dim padded to a multiple of 8).

#include "common.hip.h"
#include "kernels.h"

namespace {

constexpr int kBlock = 256;
constexpr int kFChunk = 8;

__global__ __launch_bounds__(kBlock) void gbt_histogram_k(
    const unsigned char* __restrict__ B, const float* __restrict__ g,
    const float* __restrict__ h, const int* __restrict__ row_list,
    const int4* __restrict__ bmap, int F, float* __restrict__ hist) {
  __shared__ float lh[kFChunk][256][2];  // 16 KiB
  const int4 wi = bmap[blockIdx.x];
  const int node = wi.x, f0 = wi.y, start = wi.z, count = wi.w;

  for (int i = threadIdx.x; i < kFChunk * 256 * 2; i += kBlock)
    ((float*)lh)[i] = 0.f;
  __syncthreads();

  for (int i = threadIdx.x; i < count; i += kBlock) {
    const int row = row_list[start + i];
    const unsigned long long bins8 =
        *(const unsigned long long*)(B + (long long)row * F + f0);
    const float gv = g[row];
    const float hv = h[row];
#pragma unroll
    for (int j = 0; j < kFChunk; ++j) {
      const int bin = (int)((bins8 >> (8 * j)) & 0xffu);
      atomicAdd(&lh[j][bin][0], gv);
      atomicAdd(&lh[j][bin][1], hv);
    }
  }
  __syncthreads();

  float* out = hist + ((long long)node * F + f0) * 256 * 2;
  for (int i = threadIdx.x; i < kFChunk * 256 * 2; i += kBlock)
    atomicAdd(&out[i], ((const float*)lh)[i]);
}

}  // namespace

void launch_gbt_histogram(const unsigned char* B, const float* g,
                          const float* h, const int* row_list,
                          const int4* bmap, int nblocks, int F,
                          float* hist, hipStream_t stream) {
  hipLaunchKernelGGL(gbt_histogram_k, dim3(nblocks), dim3(kBlock), 0,
                     stream, B, g, h, row_list, bmap, F, hist);
}
