// EXPERIMENTAL — schedule iteration 2 for the 256x256 bf16 GEMM.
//
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 \
//       experimental/gemm256_v2.hip -o /tmp/gemm256v2 && /tmp/gemm256v2
//
// Measured so far (gemm256_ring.hip, MI355X):
//   drain 2-phase 256x256 BK=64 + swz + setprio : 1129 TF @4096^3
//   ring  8-barrier 4-phase counted-vmcnt       : 1050 TF @4096^3
// This probe A/Bs two further schedules against those:
//
// V1 "ring1b": the ring schedule with ONE barrier per phase (no
//   mid-barrier). Safety: a phase's ds_reads complete before its own
//   MFMAs (compiler lgkmcnt), which precede the end barrier, so next
//   phase's staging cannot WAR them; RAW coverage unchanged (counted
//   vmcnt before the end barrier covers reads in later phases).
//
// V5 "bk32": BK=32, 4-deep FULL-tile ring (4 x (256x32 A + 256x32 B)
//   = 128 KiB), 2-tile prefetch, ONE barrier per K-tile, counted
//   vmcnt(4):
//     tile t: { stage(t+2) [4 calls]; ds_read 8A+4B; setprio(1);
//               32 MFMA; setprio(0); vmcnt(4); barrier }
//   Wait derivation: reads of tile t+1 (after the end-of-t barrier)
//   need stage(t+1) (issued top of t-1, 4 calls) landed; the newest 4
//   calls (stage(t+2)) may stay in flight -> vmcnt(4). WAR: buffer
//   (t+2)&3 last read during tile t-2, completed before that tile's
//   barrier, staging issued two barriers later.
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

constexpr int BM = 256, BN = 256;
constexpr int THREADS = 512;

__device__ __forceinline__ short f2bf(float f) {
  union { float f; unsigned u; } c;
  c.f = f;
  unsigned lsb = (c.u >> 16) & 1u;
  c.u += 0x7fffu + lsb;
  return (short)(c.u >> 16);
}

// ---------------------------------------------------------------- V1
// BK=64 half-tile ring, single barrier per phase.
namespace v1 {
constexpr int BK = 64;
constexpr int HALF_ELEMS = 128 * BK;

__device__ __forceinline__ int swz(int byte_off) {
  return byte_off ^ (((byte_off >> 7) & 7) << 4);
}
__device__ __forceinline__ bf16x8 ld_frag(const short* slot, int row,
                                          int kk) {
  int byte = swz((row * BK + kk) * 2);
  return *(const bf16x8*)((const char*)slot + byte);
}

__global__ __launch_bounds__(THREADS) void gemm_k(
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
    bf16x8 bfr[4][2];
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      bf16x8 afr[2][2];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int ii = 0; ii < 2; ++ii)
          afr[ii][ks] = ld_frag(sA, q * 32 + ii * 16 + frag_row,
                                ks * 32 + frag_k);
      if (q == 0) {
        const short* sB = slotB(u, wc >> 1);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            bfr[j][ks] = ld_frag(sB, (wc & 1) * 64 + j * 16 + frag_row,
                                 ks * 32 + frag_k);
      }
      if (q == 0 && u + 1 < T)
        stage_half(slotA(u + 1, 0), A, a_row0, 0, (u + 1) * BK, K);
      else if (q == 1 && u + 1 < T)
        stage_half(slotA(u + 1, 1), A, a_row0, 1, (u + 1) * BK, K);
      else if (q == 2 && u + 2 < T)
        stage_half(slotB(u + 2, 0), W, b_row0, 0, (u + 2) * BK, K);
      else if (q == 3 && u + 2 < T)
        stage_half(slotB(u + 2, 1), W, b_row0, 1, (u + 2) * BK, K);

      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int ii = 0; ii < 2; ++ii)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[q * 2 + ii][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afr[ii][ks], bfr[j][ks], acc[q * 2 + ii][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      if (q == 3) {
        if (u + 2 < T)
          asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();
    }
  }

  const int c_sub_row = (lane / 16) * 4;
  const int c_col = lane % 16;
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const long long col = b_row0 + wc * 64 + j * 16 + c_col;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const long long row = a_row0 + wr * 128 + i * 16 + c_sub_row + rr;
        C[row * N + col] = f2bf(acc[i][j][rr]);
      }
    }
}
}  // namespace v1

// ---------------------------------------------------------------- V5
// BK=32, 4-deep full-tile ring, one barrier per K-tile, vmcnt(4).
namespace v5 {
constexpr int BK = 32;
constexpr int TILE_ELEMS = 256 * BK;  // one operand tile: 16 KiB

// rows are 64 B at BK=32; ds_read_b128 of 16 rows x 16 B column is an
// 8-way conflict (2 rows/bank-pair). Swizzle: XOR byte bits 4-5 with
// row bits 0-1 (byte bits 6-7) -> 16 rows spread over 4 slots.
__device__ __forceinline__ int swz(int byte_off) {
  return byte_off ^ (((byte_off >> 6) & 3) << 4);
}
__device__ __forceinline__ bf16x8 ld_frag(const short* buf, int row,
                                          int kk) {
  int byte = swz((row * BK + kk) * 2);
  return *(const bf16x8*)((const char*)buf + byte);
}

__global__ __launch_bounds__(THREADS) void gemm_k(
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

  // buffer b (= t&3): A at lds + b*2*TILE, B at +TILE
  auto bufA = [&](int t) {
    return lds + (size_t)(t & 3) * 2 * TILE_ELEMS;
  };
  auto bufB = [&](int t) {
    return lds + (size_t)((t & 3) * 2 + 1) * TILE_ELEMS;
  };
  // stage one 256x32 operand tile: 2 calls x 512 threads x 16 B
  auto stage = [&](short* dst, const short* g, long long row0, int k0,
                   int ld) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      const int e_lin = (c * THREADS + tid) * 8;
      const int e = swz(e_lin * 2) / 2;
      const int row = e / BK, kk = e % BK;
      const short* gp = g + (row0 + row) * (long long)ld + k0 + kk;
      short* lp = dst + (c * THREADS + (tid & ~63)) * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp,
          (__attribute__((address_space(3))) unsigned int*)lp, 16, 0, 0);
    }
  };

  const int T = K / BK;
  // prologue: tiles 0 and 1 staged, full drain
  stage(bufA(0), A, a_row0, 0, K);
  stage(bufB(0), W, b_row0, 0, K);
  if (T > 1) {
    stage(bufA(1), A, a_row0, BK, K);
    stage(bufB(1), W, b_row0, BK, K);
  }
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();

  for (int t = 0; t < T; ++t) {
    if (t + 2 < T) {
      stage(bufA(t + 2), A, a_row0, (t + 2) * BK, K);
      stage(bufB(t + 2), W, b_row0, (t + 2) * BK, K);
    }
    bf16x8 a[8], b[4];
#pragma unroll
    for (int i = 0; i < 8; ++i)
      a[i] = ld_frag(bufA(t), wr * 128 + i * 16 + frag_row, frag_k);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      b[j] = ld_frag(bufB(t), wc * 64 + j * 16 + frag_row, frag_k);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[i], b[j], acc[i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    // reads of tile t+1 need stage(t+1) landed; stage(t+2)'s 4 calls
    // may stay in flight
    if (t + 2 < T)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  const int c_sub_row = (lane / 16) * 4;
  const int c_col = lane % 16;
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const long long col = b_row0 + wc * 64 + j * 16 + c_col;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const long long row = a_row0 + wr * 128 + i * 16 + c_sub_row + rr;
        C[row * N + col] = f2bf(acc[i][j][rr]);
      }
    }
}
}  // namespace v5

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

template <int V>
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
  const void* kfn = V == 1 ? (const void*)&v1::gemm_k
                           : (const void*)&v5::gemm_k;
  HIP_CHECK(hipFuncSetAttribute(
      kfn, hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds_bytes));
  auto launch = [&]() {
    if (V == 1)
      hipLaunchKernelGGL(v1::gemm_k, dim3(grid), dim3(THREADS), lds_bytes,
                         0, dA, dW, dC, M, N, K);
    else
      hipLaunchKernelGGL(v5::gemm_k, dim3(grid), dim3(THREADS), lds_bytes,
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
    printf("refcheck %dx%dx%d V%d x%d: %s (%d bad)\n", M, N, K, V,
           repeats, bad ? "FAIL" : "ok", bad);
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
    printf("perf %dx%dx%d V%d: %.3f ms/iter, %.0f TFLOP/s\n", M, N, K, V,
           ms / iters, tf);
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
    bad += run_case<1>(256, 256, 64, true, 1, 3);
    bad += run_case<1>(512, 512, 192, true, 1, 3);
    bad += run_case<1>(512, 512, 768, true, 1, 5);
    bad += run_case<5>(256, 256, 32, true, 1, 3);
    bad += run_case<5>(256, 256, 64, true, 1, 3);
    bad += run_case<5>(256, 256, 96, true, 1, 3);
    bad += run_case<5>(512, 512, 192, true, 1, 3);
    bad += run_case<5>(512, 512, 768, true, 1, 5);
    if (bad) {
      printf("REFCHECK FAILED - do not trust perf numbers\n");
      return 1;
    }
  }
  run_case<1>(4096, 4096, 4096, false, 10);
  run_case<5>(4096, 4096, 4096, false, 10);
  run_case<1>(8192, 8192, 8192, false, 5);
  run_case<5>(8192, 8192, 8192, false, 5);
  run_case<1>(32768, 3072, 768, false, 10);
  run_case<5>(32768, 3072, 768, false, 10);
  run_case<1>(32768, 768, 3072, false, 10);
  run_case<5>(32768, 768, 3072, false, 10);
  run_case<1>(32768, 768, 768, false, 10);
  run_case<5>(32768, 768, 768, false, 10);
  run_case<1>(32768, 2304, 768, false, 10);
  run_case<5>(32768, 2304, 768, false, 10);
  return 0;
}
