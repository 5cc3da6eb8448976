// Fused multi-tensor AdamW / SGD update kernels.
//
// Single-pass parameter + moment update over a chunked multi-tensor
// table: one workgroup per 16K-element chunk, float4 (16 B/lane) loads —
// the coalescing sweet spot — with a scalar tail. The whole optimizer
// step for a model is ONE kernel launch regardless of tensor count,
// replacing the per-tensor launch storm of eager optimizers.
//
// Memory-bound op: AdamW touches 4 reads + 3 writes x 4 B = 28 B/elem;
// the roofline is HBM3E (~6.3 TB/s achievable), so the design goal is
// purely maximal effective bandwidth (vector width + ≫256 workgroups).

#include "common.hip.h"
#include "kernels.h"

namespace {

constexpr int kBlock = 256;

// Advance the device step counter and publish the bias-correction
// coefficients — runs on-device so a hipGraph replay of the whole
// optimizer step still sees a fresh step number.
__global__ void adam_prep_k(long long* step_counter, float* coeffs,
                            float b1, float b2) {
  const long long step = ++step_counter[0];
  coeffs[0] = 1.f / (1.f - powf(b1, (float)step));        // 1/bc1
  coeffs[1] = rsqrtf(1.f - powf(b2, (float)step));        // 1/sqrt(bc2)
}

__global__ __launch_bounds__(kBlock) void fused_adamw_k(
    const TensorChunk* __restrict__ chunks, const int2* __restrict__ bmap,
    float lr, float b1, float b2, float eps, float wd,
    const float* __restrict__ coeffs) {
  const float inv_bc1 = coeffs[0];
  const float rsqrt_bc2 = coeffs[1];
  const int2 wi = bmap[blockIdx.x];
  const TensorChunk tc = chunks[wi.x];
  const long long start = wi.y;
  const long long end = min((long long)(start + kOptChunk), tc.n);
  const bool lowp = tc.pl != nullptr;  // wave-uniform: no divergence

  const float mb1 = 1.f - b1, mb2 = 1.f - b2;
  const float step_size = lr * inv_bc1;
  const float decay = 1.f - lr * wd;  // AdamW decoupled decay

  // float4 main body: chunk starts are 16K-aligned, so only the final
  // chunk of each tensor can have a non-multiple-of-4 tail.
  const long long vend = start + ((end - start) & ~3LL);
  for (long long i = start + (long long)threadIdx.x * 4; i < vend;
       i += (long long)kBlock * 4) {
    float4v p = *(const float4v*)(tc.p + i);
    float4v g;
    if (lowp) {
      const short4v gs = *(const short4v*)((const short*)tc.g + i);
#pragma unroll
      for (int j = 0; j < 4; ++j) g[j] = bf2f(gs[j]);
    } else {
      g = *(const float4v*)(tc.g + i);
    }
    float4v m = *(const float4v*)(tc.m + i);
    float4v v = *(const float4v*)(tc.v + i);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      m[j] = b1 * m[j] + mb1 * g[j];
      v[j] = b2 * v[j] + mb2 * g[j] * g[j];
      p[j] = decay * p[j] -
             step_size * m[j] / (sqrtf(v[j]) * rsqrt_bc2 + eps);
    }
    *(float4v*)(tc.m + i) = m;
    *(float4v*)(tc.v + i) = v;
    *(float4v*)(tc.p + i) = p;
    if (lowp) {
      short4v ps;
#pragma unroll
      for (int j = 0; j < 4; ++j) ps[j] = f2bf(p[j]);
      *(short4v*)(tc.pl + i) = ps;
    }
  }
  for (long long i = vend + threadIdx.x; i < end; i += kBlock) {
    const float g = lowp ? bf2f(((const short*)tc.g)[i]) : tc.g[i];
    const float m = b1 * tc.m[i] + mb1 * g;
    const float v = b2 * tc.v[i] + mb2 * g * g;
    tc.m[i] = m;
    tc.v[i] = v;
    const float p =
        decay * tc.p[i] - step_size * m / (sqrtf(v) * rsqrt_bc2 + eps);
    tc.p[i] = p;
    if (lowp) tc.pl[i] = f2bf(p);
  }
}

__global__ __launch_bounds__(kBlock) void fused_sgd_k(
    const TensorChunk* __restrict__ chunks, const int2* __restrict__ bmap,
    float lr, float momentum, float wd, int nesterov, int first_step) {
  const int2 wi = bmap[blockIdx.x];
  const TensorChunk tc = chunks[wi.x];
  const long long start = wi.y;
  const long long end = min((long long)(start + kOptChunk), tc.n);

  const long long vend = start + ((end - start) & ~3LL);
  for (long long i = start + (long long)threadIdx.x * 4; i < vend;
       i += (long long)kBlock * 4) {
    float4v p = *(const float4v*)(tc.p + i);
    float4v g = *(const float4v*)(tc.g + i);
    float4v mu = *(const float4v*)(tc.m + i);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      g[j] += wd * p[j];
      mu[j] = first_step ? g[j] : momentum * mu[j] + g[j];
      const float upd = nesterov ? g[j] + momentum * mu[j] : mu[j];
      p[j] -= lr * upd;
    }
    *(float4v*)(tc.m + i) = mu;
    *(float4v*)(tc.p + i) = p;
  }
  for (long long i = vend + threadIdx.x; i < end; i += kBlock) {
    float g = tc.g[i] + wd * tc.p[i];
    const float mu = first_step ? g : momentum * tc.m[i] + g;
    tc.m[i] = mu;
    tc.p[i] -= lr * (nesterov ? g + momentum * mu : mu);
  }
}

__global__ __launch_bounds__(kBlock) void zero_grads_k(
    const TensorChunk* __restrict__ chunks,
    const int2* __restrict__ bmap) {
  const int2 wi = bmap[blockIdx.x];
  const TensorChunk tc = chunks[wi.x];
  const long long start = wi.y;
  const long long end = min((long long)(start + kOptChunk), tc.n);
  const long long vend = start + ((end - start) & ~3LL);
  if (tc.pl != nullptr) {  // bf16 grads
    short* g = const_cast<short*>((const short*)tc.g);
    const short4v z = {0, 0, 0, 0};
    for (long long i = start + (long long)threadIdx.x * 4; i < vend;
         i += (long long)kBlock * 4)
      *(short4v*)(g + i) = z;
    for (long long i = vend + threadIdx.x; i < end; i += kBlock) g[i] = 0;
  } else {
    float* g = const_cast<float*>(tc.g);
    const float4v z = {0.f, 0.f, 0.f, 0.f};
    for (long long i = start + (long long)threadIdx.x * 4; i < vend;
         i += (long long)kBlock * 4)
      *(float4v*)(g + i) = z;
    for (long long i = vend + threadIdx.x; i < end; i += kBlock)
      g[i] = 0.f;
  }
}

}  // namespace

void launch_zero_grads(const TensorChunk* chunks, const int2* bmap,
                       int nblocks, hipStream_t stream) {
  hipLaunchKernelGGL(zero_grads_k, dim3(nblocks), dim3(kBlock), 0, stream,
                     chunks, bmap);
}

void launch_fused_adamw(const TensorChunk* chunks, const int2* bmap,
                        int nblocks, float lr, float beta1, float beta2,
                        float eps, float weight_decay,
                        long long* step_counter, float* coeffs,
                        hipStream_t stream) {
  hipLaunchKernelGGL(adam_prep_k, dim3(1), dim3(1), 0, stream,
                     step_counter, coeffs, beta1, beta2);
  hipLaunchKernelGGL(fused_adamw_k, dim3(nblocks), dim3(kBlock), 0, stream,
                     chunks, bmap, lr, beta1, beta2, eps, weight_decay,
                     coeffs);
}

void launch_fused_sgd(const TensorChunk* chunks, const int2* bmap,
                      int nblocks, float lr, float momentum,
                      float weight_decay, bool nesterov, bool first_step,
                      hipStream_t stream) {
  hipLaunchKernelGGL(fused_sgd_k, dim3(nblocks), dim3(kBlock), 0, stream,
                     chunks, bmap, lr, momentum, weight_decay,
                     (int)nesterov, (int)first_step);
}
