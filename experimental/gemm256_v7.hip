// EXPERIMENTAL — G19: de-phased pairs of 8-wave workgroups.
//
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 \
//       experimental/gemm256_v7.hip -o /tmp/g7 && /tmp/g7
//
// G16 (16-wave 256x256, 1 WG/CU) reaches 1201 TF @4k; its barriers
// synchronize ALL 4 waves/SIMD so every wave stalls together at the
// per-tile drain. This variant keeps 4 waves/SIMD but splits them into
// TWO independent 8-wave workgroups (256x128 tile, BK=32, 48 KiB LDS
// each -> 2 WGs/CU by LDS, 4 waves/SIMD by VGPR): the two WGs' drains
// interleave, so one WG's MFMA covers the other's staging stall.
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define HIP_CHECK(x)                                                     \
  do {                                                                   \
    hipError_t e = (x);                                                  \
    if (e != hipSuccess) {                                               \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(e), __FILE__,  \
             __LINE__);                                                  \
      exit(1);                                                           \
    }                                                                    \
  } while (0)

constexpr int BM = 256, BN = 128, BK = 32;
constexpr int THREADS = 512;  // 8 waves: 4 (M) x 2 (N)

__device__ __forceinline__ short f2bf(float f) {
  union { float f; unsigned u; } c;
  c.f = f;
  unsigned lsb = (c.u >> 16) & 1u;
  c.u += 0x7fffu + lsb;
  return (short)(c.u >> 16);
}
// rows are 64 B at BK=32: XOR byte bits 4-5 with row bits 0-1
__device__ __forceinline__ int swz(int byte_off) {
  return byte_off ^ (((byte_off >> 6) & 3) << 4);
}
__device__ __forceinline__ bf16x8 ld_frag(const short* buf, int row,
                                          int kk) {
  return *(const bf16x8*)((const char*)buf + swz((row * BK + kk) * 2));
}

__global__ __launch_bounds__(THREADS) void gemm_g19_k(
    const short* __restrict__ A, const short* __restrict__ W,
    short* __restrict__ C, int M, int N, int K) {
  extern __shared__ short lds[];
  const int nwg = gridDim.x;
  const int q_ = nwg / 8, r_ = nwg % 8;
  const int xcd = blockIdx.x % 8, idx = blockIdx.x / 8;
  const int wgid =
      (xcd < r_ ? xcd * (q_ + 1) : r_ * (q_ + 1) + (xcd - r_) * q_) + idx;
  const int ntn = N / BN;
  const long long a_row0 = (long long)(wgid / ntn) * BM;
  const long long b_row0 = (long long)(wgid % ntn) * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;  // 0..3 -> rows [wr*64, +64)
  const int wc = wave & 1;   // 0..1 -> cols [wc*64, +64)
  const int frag_row = lane % 16;
  const int frag_k = (lane / 16) * 8;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // buffers: A 256x32 (16 KiB) + B 128x32 (8 KiB), double-buffered
  auto bufA = [&](int b) { return lds + (size_t)b * (BM + BN) * BK; };
  auto bufB = [&](int b) {
    return lds + (size_t)b * (BM + BN) * BK + BM * BK;
  };
  // stage: A tile 8192 elems (2 calls x 512thr x 8), B tile 4096 (1)
  auto stage = [&](short* la, short* lb, long long k0) {
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      const int e0 = (s * THREADS + tid) * 8;
      const int e = swz(e0 * 2) / 2;
      const short* gp = A + (a_row0 + e / BK) * (long long)K + k0 + e % BK;
      short* lp = la + ((s * THREADS + (tid & ~63)) * 8);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp,
          (__attribute__((address_space(3))) unsigned int*)lp, 16, 0, 0);
    }
    {
      const int e0 = tid * 8;
      const int e = swz(e0 * 2) / 2;
      const short* gp = W + (b_row0 + e / BK) * (long long)K + k0 + e % BK;
      short* lp = lb + ((tid & ~63) * 8);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp,
          (__attribute__((address_space(3))) unsigned int*)lp, 16, 0, 0);
    }
  };

  stage(bufA(0), bufB(0), 0);
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();

  int cur = 0;
  const int ntiles = K / BK;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles)
      stage(bufA(cur ^ 1), bufB(cur ^ 1), (long long)(t + 1) * BK);
    bf16x8 a[4], b[4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
      a[i] = ld_frag(bufA(cur), wr * 64 + i * 16 + frag_row, frag_k);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      b[j] = ld_frag(bufB(cur), wc * 64 + j * 16 + frag_row, frag_k);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[i], b[j], acc[i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_waitcnt(0);
    __syncthreads();
    cur ^= 1;
  }

  const int c_sub_row = (lane / 16) * 4;
  const int c_col = lane % 16;
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const long long col = b_row0 + wc * 64 + j * 16 + c_col;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const long long row = a_row0 + wr * 64 + i * 16 + c_sub_row + rr;
        C[row * N + col] = f2bf(acc[i][j][rr]);
      }
    }
}

static void cpu_ref(const std::vector<short>& A, const std::vector<short>& W,
                    std::vector<float>& C, int M, int N, int K) {
  auto b2f = [](short s) {
    union { float f; unsigned u; } c;
    c.u = ((unsigned)(unsigned short)s) << 16;
    return c.f;
  };
  for (int m = 0; m < M; ++m)
    for (int n = 0; n < N; ++n) {
      float acc = 0.f;
      for (int k = 0; k < K; ++k)
        acc += b2f(A[(size_t)m * K + k]) * b2f(W[(size_t)n * K + k]);
      C[(size_t)m * N + n] = acc;
    }
}
static short host_f2bf(float f) {
  union { float f; unsigned u; } c;
  c.f = f;
  unsigned lsb = (c.u >> 16) & 1u;
  c.u += 0x7fffu + lsb;
  return (short)(c.u >> 16);
}

static int run_case(int M, int N, int K, bool check, int iters,
                    int repeats = 1) {
  std::vector<short> hA((size_t)M * K), hW((size_t)N * K);
  srand(42);
  for (auto& v : hA) v = host_f2bf((rand() % 2000 - 1000) / 500.0f);
  for (auto& v : hW) v = host_f2bf((rand() % 2000 - 1000) / 500.0f);
  short *dA, *dW, *dC;
  HIP_CHECK(hipMalloc(&dA, hA.size() * 2));
  HIP_CHECK(hipMalloc(&dW, hW.size() * 2));
  HIP_CHECK(hipMalloc(&dC, (size_t)M * N * 2));
  HIP_CHECK(hipMemcpy(dA, hA.data(), hA.size() * 2, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(dW, hW.data(), hW.size() * 2, hipMemcpyHostToDevice));
  const int grid = (M / BM) * (N / BN);
  const size_t lds_bytes = 2 * (size_t)(BM + BN) * BK * sizeof(short);
  HIP_CHECK(hipFuncSetAttribute(
      (const void*)&gemm_g19_k, hipFuncAttributeMaxDynamicSharedMemorySize,
      (int)lds_bytes));
  auto launch = [&]() {
    hipLaunchKernelGGL(gemm_g19_k, dim3(grid), dim3(THREADS), lds_bytes,
                       0, dA, dW, dC, M, N, K);
  };
  int bad = 0;
  if (check) {
    std::vector<float> ref((size_t)M * N);
    cpu_ref(hA, hW, ref, M, N, K);
    std::vector<short> hC((size_t)M * N);
    for (int rep = 0; rep < repeats; ++rep) {
      HIP_CHECK(hipMemset(dC, 0, (size_t)M * N * 2));
      launch();
      HIP_CHECK(hipDeviceSynchronize());
      HIP_CHECK(hipMemcpy(hC.data(), dC, hC.size() * 2,
                          hipMemcpyDeviceToHost));
      int rb = 0;
      for (size_t i = 0; i < hC.size(); ++i) {
        union { float f; unsigned u; } c;
        c.u = ((unsigned)(unsigned short)hC[i]) << 16;
        if (fabsf(c.f - ref[i]) > 2e-2f + 2e-2f * fabsf(ref[i])) {
          if (rb < 3) printf("  mismatch [%zu]: %f vs %f\n", i, c.f, ref[i]);
          ++rb;
        }
      }
      bad += rb;
    }
    printf("refcheck %dx%dx%d g19 x%d: %s (%d bad)\n", M, N, K, repeats,
           bad ? "FAIL" : "ok", bad);
  } else {
    launch();
    HIP_CHECK(hipDeviceSynchronize());
    hipEvent_t e0, e1;
    HIP_CHECK(hipEventCreate(&e0));
    HIP_CHECK(hipEventCreate(&e1));
    HIP_CHECK(hipEventRecord(e0));
    for (int it = 0; it < iters; ++it) launch();
    HIP_CHECK(hipEventRecord(e1));
    HIP_CHECK(hipEventSynchronize(e1));
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
    printf("perf %dx%dx%d g19: %.3f ms/iter, %.0f TFLOP/s\n", M, N, K,
           ms / iters, 2.0 * M * N * K * iters / (ms / 1e3) / 1e12);
  }
  HIP_CHECK(hipFree(dA));
  HIP_CHECK(hipFree(dW));
  HIP_CHECK(hipFree(dC));
  return bad;
}

int main() {
  int bad = 0;
  bad += run_case(256, 128, 64, true, 1, 3);
  bad += run_case(512, 256, 192, true, 1, 3);
  bad += run_case(512, 512, 768, true, 1, 5);
  if (bad) {
    printf("REFCHECK FAILED\n");
    return 1;
  }
  run_case(4096, 4096, 4096, false, 10);
  run_case(8192, 8192, 8192, false, 5);
  run_case(32768, 3072, 768, false, 10);
  run_case(32768, 768, 3072, false, 10);
  run_case(32768, 768, 768, false, 10);
  run_case(32768, 2304, 768, false, 10);
  return 0;
}
