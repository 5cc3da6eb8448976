// EXPERIMENTAL — schedule iteration 3 for the 256x256 bf16 GEMM.
//
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 \
//       experimental/gemm256_v3.hip -o /tmp/g3 && /tmp/g3
//
// Measured so far @4096^3 (MI355X): drain=1129, ring(4ph,2bar)=1050,
// ring(4ph,1bar)=1109, bk32=987; hipBLASLt=1450.
//
// V8: 2 phases per K-tile, counted vmcnt, half-tile rings (BK=64).
//   phase a: { ds_read B panel (8) + A quadrants 0-1 (8);
//              stage A0(u+1), A1(u+1); lgkmcnt(0); setprio;
//              32 MFMA; barrier }
//   phase b: { ds_read A quadrants 2-3 (8);
//              stage B0(u+2), B1(u+2); lgkmcnt(0); setprio;
//              32 MFMA; vmcnt(4)|vmcnt(0); barrier }
//   Safety: B(u) frags are ds_read in phase a, complete before the
//   phase-a barrier (lgkmcnt before MFMA); B(u+2) staging into the
//   same slots issues in phase b — WAR safe. A(u+1) staged phase a,
//   first read tile u+1 phase a, covered by the vmcnt(4) before the
//   phase-b barrier (allows only B(u+2)'s 4 calls outstanding).
//   Bigger MFMA clusters (32) than the 4-phase ring, half the
//   barriers, and no vmcnt(0) drain (the drain kernel's stall).
//
// V8e: V8 + LDS-staged coalesced C writeout. The direct writeout
//   stores per-lane columns (4 consecutive rows x 1 col -> 16-bit
//   scattered stores). V8e stages the 256x256 bf16 output tile in
//   LDS (128 KiB, free after the K-loop) and streams it out as
//   bf16x8 row-contiguous stores (fully coalesced). Matters most at
//   small K where C traffic ~ A traffic.
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
constexpr int THREADS = 512;
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

template <bool LDSOUT>
__global__ __launch_bounds__(THREADS) void gemm_v8_k(
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

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  auto slotA = [&](int t, int h) {
    return lds + (size_t)((2 * t + h) & 3) * HALF_ELEMS;
  };
  auto slotB = [&](int t, int h) {
    return lds + (size_t)(4 + ((2 * t + h) & 3)) * HALF_ELEMS;
  };
  auto stage_half = [&](short* slot, const short* g, long long row0,
                        int h, int k0, int ld) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      const int e_lin = (c * THREADS + tid) * 8;
      const int e = swz(e_lin * 2) / 2;
      const int row = e / BK, kk = e % BK;
      const short* gp =
          g + (row0 + h * 128 + row) * (long long)ld + k0 + kk;
      short* lp = slot + (c * THREADS + (tid & ~63)) * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp,
          (__attribute__((address_space(3))) unsigned int*)lp, 16, 0, 0);
    }
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
    const short* sA = slotA(u, wr);
    const short* sB = slotB(u, wc >> 1);
    // ---- phase a: B panel + A quadrants 0-1 ----
    bf16x8 bfr[4][2], afr[4][2];
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
      for (int j = 0; j < 4; ++j)
        bfr[j][ks] = ld_frag(sB, (wc & 1) * 64 + j * 16 + frag_row,
                             ks * 32 + frag_k);
#pragma unroll
      for (int ii = 0; ii < 4; ++ii)
        afr[ii][ks] =
            ld_frag(sA, ii * 16 + frag_row, ks * 32 + frag_k);
    }
    if (u + 1 < T) {
      stage_half(slotA(u + 1, 0), A, a_row0, 0, (u + 1) * BK, K);
      stage_half(slotA(u + 1, 1), A, a_row0, 1, (u + 1) * BK, K);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
#pragma unroll
      for (int ii = 0; ii < 4; ++ii)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[ii][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[ii][ks], bfr[j][ks], acc[ii][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ---- phase b: A quadrants 2-3 ----
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
#pragma unroll
      for (int ii = 0; ii < 4; ++ii)
        afr[ii][ks] =
            ld_frag(sA, 64 + ii * 16 + frag_row, ks * 32 + frag_k);
    if (u + 2 < T) {
      stage_half(slotB(u + 2, 0), W, b_row0, 0, (u + 2) * BK, K);
      stage_half(slotB(u + 2, 1), W, b_row0, 1, (u + 2) * BK, K);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
#pragma unroll
      for (int ii = 0; ii < 4; ++ii)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[4 + ii][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[ii][ks], bfr[j][ks], acc[4 + ii][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    if (u + 2 < T)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  if (!LDSOUT) {
    const int c_sub_row = (lane / 16) * 4;
    const int c_col = lane % 16;
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const long long col = b_row0 + wc * 64 + j * 16 + c_col;
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          const long long row =
              a_row0 + wr * 128 + i * 16 + c_sub_row + rr;
          C[row * N + col] = f2bf(acc[i][j][rr]);
        }
      }
    return;
  }

  // LDS-staged coalesced writeout: stage the 256x256 bf16 output tile
  // ([row][256], 128 KiB) and stream it out row-contiguously.
  __builtin_amdgcn_s_barrier();  // all K-loop LDS traffic complete
  {
    const int c_sub_row = (lane / 16) * 4;
    const int c_col = lane % 16;
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int col = wc * 64 + j * 16 + c_col;
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          const int row = wr * 128 + i * 16 + c_sub_row + rr;
          lds[row * BN + col] = f2bf(acc[i][j][rr]);
        }
      }
  }
  __builtin_amdgcn_s_barrier();
  // 512 threads x 16 B = 16 rows per sweep; 16 sweeps
#pragma unroll
  for (int s = 0; s < 16; ++s) {
    const int row = s * 16 + tid / 32;
    const int col = (tid % 32) * 8;
    const bf16x8 v = *(const bf16x8*)&lds[row * BN + col];
    *(bf16x8*)&C[(a_row0 + row) * (long long)N + b_row0 + col] = v;
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

template <bool LDSOUT>
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
  const size_t lds_bytes = 128 * 1024;
  HIP_CHECK(hipFuncSetAttribute(
      (const void*)&gemm_v8_k<LDSOUT>,
      hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds_bytes));
  auto launch = [&]() {
    hipLaunchKernelGGL(gemm_v8_k<LDSOUT>, dim3(grid), dim3(THREADS),
                       lds_bytes, 0, dA, dW, dC, M, N, K);
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
    printf("refcheck %dx%dx%d ldsout=%d x%d: %s (%d bad)\n", M, N, K,
           (int)LDSOUT, repeats, bad ? "FAIL" : "ok", bad);
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
    printf("perf %dx%dx%d ldsout=%d: %.3f ms/iter, %.0f TFLOP/s\n", M, N,
           K, (int)LDSOUT, ms / iters, tf);
  }
  HIP_CHECK(hipFree(dA));
  HIP_CHECK(hipFree(dW));
  HIP_CHECK(hipFree(dC));
  return bad;
}

int main(int argc, char** argv) {
  const bool perf_only = argc > 1 && argv[1][0] == 'p';
  int bad = 0;
  if (!perf_only) {
    bad += run_case<false>(256, 256, 64, true, 1, 3);
    bad += run_case<false>(512, 512, 192, true, 1, 3);
    bad += run_case<false>(512, 512, 768, true, 1, 5);
    bad += run_case<true>(256, 256, 64, true, 1, 3);
    bad += run_case<true>(512, 512, 192, true, 1, 3);
    bad += run_case<true>(512, 512, 768, true, 1, 5);
    if (bad) {
      printf("REFCHECK FAILED - do not trust perf numbers\n");
      return 1;
    }
  }
  run_case<false>(4096, 4096, 4096, false, 10);
  run_case<true>(4096, 4096, 4096, false, 10);
  run_case<false>(8192, 8192, 8192, false, 5);
  run_case<true>(8192, 8192, 8192, false, 5);
  run_case<false>(32768, 3072, 768, false, 10);
  run_case<true>(32768, 3072, 768, false, 10);
  run_case<false>(32768, 768, 3072, false, 10);
  run_case<true>(32768, 768, 3072, false, 10);
  run_case<false>(32768, 768, 768, false, 10);
  run_case<true>(32768, 768, 768, false, 10);
  run_case<false>(32768, 2304, 768, false, 10);
  run_case<true>(32768, 2304, 768, false, 10);
  return 0;
}
