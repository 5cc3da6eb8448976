// EXPERIMENTAL — G17: 16-wave 256x256 BK=64 GEMM with the counted-
// vmcnt half-tile ring (v5's G16 occupancy + v3's V8 schedule).
//
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 \
//       experimental/gemm256_v6.hip -o /tmp/g6 && /tmp/g6
//
// Measured context: 8-wave designs cap at 2 waves/SIMD (214+ VGPR) and
// plateau ~1100 TF; 16-wave G16 (128 VGPR -> 4 waves/SIMD) with a
// plain drain schedule reaches 1201 @4k. MFMA-only ceiling at 4
// waves/SIMD: 2389 TF. This variant removes G16's vmcnt(0) drain:
//
// Geometry: 16 waves (4x4), per-wave output 64x64 (acc 4x4), per
// K-tile 32 MFMA/wave in 2 phases of 16 (phase a = K-step 0 + B panel
// reads, phase b = K-step 1).
// LDS: A ring 4 half-slots + B ring 6 half-slots (128x64 each),
// 160 KiB.  Staging: phase a: A0(u+1), A1(u+1); phase b: B0(u+2),
// B1(u+2). 1024 threads x 16 B = one half-tile per call.
//
// Wait derivation (identical shape to v3's V8):
//   reads of tile u+1 phase a touch A(u+1) (staged tile u phase a)
//   and B(u+1) (staged tile u-1 phase b). Every wave must cover those
//   with a vmcnt wait before the end-of-tile barrier: at the phase-b
//   wait the queue (newest first) is [B1(u+2), B0(u+2) | A1(u+1),
//   A0(u+1), ...] -> s_waitcnt vmcnt(2) allows only the two B(u+2)
//   calls to stay in flight (one call per half-tile at 1024 threads).
//   WAR: A(u+1) slots last read tile u-1 phase b; B(u+2) slots last
//   read tile u phase a — both barrier-separated from the staging
//   issue.
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

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int THREADS = 1024;
constexpr int HALF_ELEMS = 128 * BK;

__device__ __forceinline__ short f2bf(float f) {
  union { float f; unsigned u; } c;
  c.f = f;
  unsigned lsb = (c.u >> 16) & 1u;
  c.u += 0x7fffu + lsb;
  return (short)(c.u >> 16);
}
__device__ __forceinline__ int swz(int byte_off) {
  return byte_off ^ (((byte_off >> 7) & 7) << 4);
}
__device__ __forceinline__ bf16x8 ld_frag(const short* slot, int row,
                                          int kk) {
  int byte = swz((row * BK + kk) * 2);
  return *(const bf16x8*)((const char*)slot + byte);
}

__global__ __launch_bounds__(THREADS) void gemm_g17_k(
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
  const int wr = wave >> 2;
  const int wc = wave & 3;
  const int frag_row = lane % 16;
  const int frag_k = (lane / 16) * 8;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  auto slotA = [&](int t, int h) {
    return lds + (size_t)((2 * t + h) & 3) * HALF_ELEMS;
  };
  auto slotB = [&](int t, int h) {
    // 6-deep ring: B is read in BOTH phases of its tile, so a 2-tile
    // prefetch distance needs tenant spacing 3 tiles (mod-6 on half
    // index) to keep staging WAR-safe (tenant B(u-1) last read at
    // tile u-1 phase b; staging issues tile u phase b).
    return lds + (size_t)(4 + ((2 * t + h) % 6)) * HALF_ELEMS;
  };
  // one half-tile (128x64) = 1024 threads x 16 B = one call
  auto stage_half = [&](short* slot, const short* g, long long row0,
                        int h, int k0, int ld) {
    const int e_lin = tid * 8;
    const int e = swz(e_lin * 2) / 2;
    const int row = e / BK, kk = e % BK;
    const short* gp = g + (row0 + h * 128 + row) * (long long)ld + k0 + kk;
    short* lp = slot + (tid & ~63) * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)gp,
        (__attribute__((address_space(3))) unsigned int*)lp, 16, 0, 0);
  };

  const int T = K / BK;
  stage_half(slotA(0, 0), A, a_row0, 0, 0, K);
  stage_half(slotA(0, 1), A, a_row0, 1, 0, K);
  stage_half(slotB(0, 0), W, b_row0, 0, 0, K);
  stage_half(slotB(0, 1), W, b_row0, 1, 0, K);
  if (T > 1) {
    stage_half(slotB(1, 0), W, b_row0, 0, BK, K);
    stage_half(slotB(1, 1), W, b_row0, 1, BK, K);
  }
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();

  for (int u = 0; u < T; ++u) {
    const short* sA = slotA(u, wr >> 1);
    const short* sB = slotB(u, wc >> 1);
    const int arow0 = (wr & 1) * 64;  // wave's rows within its A half
    // ---- phase a: K-step 0 ----
    bf16x8 bfr[4], a0[4];
#pragma unroll
    for (int j = 0; j < 4; ++j)
      bfr[j] = ld_frag(sB, (wc & 1) * 64 + j * 16 + frag_row, frag_k);
#pragma unroll
    for (int i = 0; i < 4; ++i)
      a0[i] = ld_frag(sA, arow0 + i * 16 + frag_row, frag_k);
    if (u + 1 < T) {
      stage_half(slotA(u + 1, 0), A, a_row0, 0, (u + 1) * BK, K);
      stage_half(slotA(u + 1, 1), A, a_row0, 1, (u + 1) * BK, K);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a0[i], bfr[j], acc[i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ---- phase b: K-step 1 ----
#pragma unroll
    for (int j = 0; j < 4; ++j)
      bfr[j] = ld_frag(sB, (wc & 1) * 64 + j * 16 + frag_row,
                       32 + frag_k);
#pragma unroll
    for (int i = 0; i < 4; ++i)
      a0[i] = ld_frag(sA, arow0 + i * 16 + frag_row, 32 + frag_k);
    if (u + 2 < T) {
      stage_half(slotB(u + 2, 0), W, b_row0, 0, (u + 2) * BK, K);
      stage_half(slotB(u + 2, 1), W, b_row0, 1, (u + 2) * BK, K);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a0[i], bfr[j], acc[i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    if (u + 2 < T)
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
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

// ------------------------------------------------------------ harness
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
  const size_t lds_bytes = 160 * 1024;
  HIP_CHECK(hipFuncSetAttribute(
      (const void*)&gemm_g17_k,
      hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds_bytes));
  auto launch = [&]() {
    hipLaunchKernelGGL(gemm_g17_k, dim3(grid), dim3(THREADS), lds_bytes,
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
      int rep_bad = 0;
      for (size_t i = 0; i < hC.size(); ++i) {
        union { float f; unsigned u; } c;
        c.u = ((unsigned)(unsigned short)hC[i]) << 16;
        const float got = c.f, want = ref[i];
        if (fabsf(got - want) > 2e-2f + 2e-2f * fabsf(want)) {
          if (rep_bad < 3)
            printf("  rep%d mismatch [%zu]: got %f want %f\n", rep, i,
                   got, want);
          ++rep_bad;
        }
      }
      bad += rep_bad;
    }
    printf("refcheck %dx%dx%d g17 x%d: %s (%d bad)\n", M, N, K, repeats,
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
    const double tf = 2.0 * M * N * K * iters / (ms / 1e3) / 1e12;
    printf("perf %dx%dx%d g17: %.3f ms/iter, %.0f TFLOP/s\n", M, N, K,
           ms / iters, tf);
  }
  HIP_CHECK(hipFree(dA));
  HIP_CHECK(hipFree(dW));
  HIP_CHECK(hipFree(dC));
  return bad;
}

int main() {
  int bad = 0;
  bad += run_case(256, 256, 64, true, 1, 3);
  bad += run_case(256, 256, 128, true, 1, 3);
  bad += run_case(512, 512, 192, true, 1, 3);
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
