// EXPERIMENTAL — schedule iteration 4: register-prefetch pipeline.
//
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 \
//       experimental/gemm256_v4.hip -o /tmp/g4 && /tmp/g4
//
// Diagnosis after v1-v3 (all ~1050-1130 TF @4096^3 vs hipBLASLt 1450):
// every variant issues its ds_reads and then waits lgkmcnt(0) right
// before the MFMAs that consume them — the LDS round-trip latency sits
// exposed on the critical path once per phase. This variant pipelines
// REGISTER loads one phase ahead: phase q issues the ds_reads for
// phase q+1's fragments and waits only on the PREVIOUS phase's reads
// (counted lgkmcnt), so LDS latency hides under 16 MFMA of work.
//
// V11 geometry: 256x256, BK=64, 8 waves (2Mx4N), 4 phases/K-tile.
//   A ring: 6 half-tile slots (3 K-tiles deep), 96 KiB
//   B ring: 4 half-tile slots (2 K-tiles deep), 64 KiB   -> 160 KiB LDS
// Phase q of tile u:
//   ds_read A-quad(q+1) frags [4 reads]  (q=3: A-quad0(u+1) + B(u+1)
//                                         panel = 12 reads)
//   stage: ph0:A0(u+2) ph1:A1(u+2) ph2:B0(u+2) ph3:B1(u+2)
//   s_waitcnt lgkmcnt(<just-issued>)   # waits only the PREVIOUS reads
//   setprio(1); 16 MFMA (quad q); setprio(0)
//   ph2 only: s_waitcnt vmcnt(6)       # see derivation
//   s_barrier
//
// WAIT DERIVATION (cross-wave, one counted vmcnt per K-tile):
//   The ph3 ds_reads touch A(u+1) (staged at tile u-1 ph0,1) and
//   B(u+1) (staged at tile u-1 ph2,3). Cross-wave safety requires
//   every wave to have covered those stagings with a vmcnt wait before
//   a barrier that precedes ph3 — i.e. by the end of ph2 of tile u.
//   Queue at that point (newest first):
//     [B0(u+2)@ph2, A1(u+2)@ph1, A0(u+2)@ph0 | B1(u+1), B0(u+1), ...]
//   Allowing the 3 newest half-tiles (2 calls each) to stay in flight:
//     s_waitcnt vmcnt(6)
//   Slack: A(u+2) staged ph0/ph1 of tile u, covered end of ph2 of
//   tile u+1 (6-7 phases); B(u+2) staged ph2/ph3, covered 3-4 phases
//   later. A-ring WAR: slot of A_h(u+2) last REG-read at tile u-1
//   (quad reads), staged at tile u ph0/1 — barrier-separated. B-ring
//   WAR: B(u+2) slot = B(u)'s, last read ph3 of tile u-1, staged
//   tile u ph2 — barrier-separated.
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

__global__ __launch_bounds__(THREADS) void gemm_v11_k(
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

  // A ring: 6 half-slots (mod-6 on half index 2t+h); B ring: 4 slots.
  auto slotA = [&](int t, int h) {
    return lds + (size_t)((2 * t + h) % 6) * HALF_ELEMS;
  };
  auto slotB = [&](int t, int h) {
    return lds + (size_t)(6 + ((2 * t + h) & 3)) * HALF_ELEMS;
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
  // prologue: A(0), A(1), B(0), B(1); drain; pre-read quad0(0) + B(0)
  stage_half(slotA(0, 0), A, a_row0, 0, 0, K);
  stage_half(slotA(0, 1), A, a_row0, 1, 0, K);
  stage_half(slotB(0, 0), W, b_row0, 0, 0, K);
  stage_half(slotB(0, 1), W, b_row0, 1, 0, K);
  if (T > 1) {
    stage_half(slotA(1, 0), A, a_row0, 0, BK, K);
    stage_half(slotA(1, 1), A, a_row0, 1, BK, K);
    stage_half(slotB(1, 0), W, b_row0, 0, BK, K);
    stage_half(slotB(1, 1), W, b_row0, 1, BK, K);
  }
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();

  bf16x8 bfr[4][2];             // B panel of the current tile (JIT)
  bf16x8 afr[2][2], anx[2][2];  // A quad: current phase, next phase
#pragma unroll
  for (int ks = 0; ks < 2; ++ks)
#pragma unroll
    for (int ii = 0; ii < 2; ++ii)
      afr[ii][ks] = ld_frag(slotA(0, wr), ii * 16 + frag_row,
                            ks * 32 + frag_k);

  for (int u = 0; u < T; ++u) {
    const short* sA = slotA(u, wr);
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      // --- issue reads: B panel JIT at ph0 (consumed this phase, so
      // the lgkm wait below must cover it), next-phase A quad
      // otherwise (ph3 reads tile u+1's quad0 — its staging was
      // covered by the previous tile's vmcnt wait) ---
      int just_issued;
      if (q == 0) {
        const short* sB = slotB(u, wc >> 1);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            bfr[j][ks] = ld_frag(sB, (wc & 1) * 64 + j * 16 + frag_row,
                                 ks * 32 + frag_k);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int ii = 0; ii < 2; ++ii)
            anx[ii][ks] = ld_frag(sA, 32 + ii * 16 + frag_row,
                                  ks * 32 + frag_k);
        just_issued = 4;  // only the A-quad1 reads may stay in flight
      } else if (q < 3) {
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int ii = 0; ii < 2; ++ii)
            anx[ii][ks] =
                ld_frag(sA, (q + 1) * 32 + ii * 16 + frag_row,
                        ks * 32 + frag_k);
        just_issued = 4;
      } else if (u + 1 < T) {
        const short* nA = slotA(u + 1, wr);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int ii = 0; ii < 2; ++ii)
            anx[ii][ks] = ld_frag(nA, ii * 16 + frag_row,
                                  ks * 32 + frag_k);
        just_issued = 4;
      } else {
        just_issued = 0;
      }

      // --- issue this phase's staging ---
      if (q == 0 && u + 2 < T)
        stage_half(slotA(u + 2, 0), A, a_row0, 0, (u + 2) * BK, K);
      else if (q == 1 && u + 2 < T)
        stage_half(slotA(u + 2, 1), A, a_row0, 1, (u + 2) * BK, K);
      else if (q == 2 && u + 2 < T)
        stage_half(slotB(u + 2, 0), W, b_row0, 0, (u + 2) * BK, K);
      else if (q == 3 && u + 2 < T)
        stage_half(slotB(u + 2, 1), W, b_row0, 1, (u + 2) * BK, K);

      // --- wait: previous phase's A reads have landed; allow only
      // this phase's 4 A-prefetch reads to stay in flight (at ph0
      // this also waits the just-issued B panel, which the MFMAs
      // below consume) ---
      if (just_issued == 4)
        asm volatile("s_waitcnt lgkmcnt(4)" ::: "memory");
      else
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

      if (q == 2) {
        // counted vmcnt (see header): A(u+1)/B(u+1) must be landed
        // before the ph3 reads; 3 newest half-tiles may stay in
        // flight.
        if (u + 2 < T)
          asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();

      // rotate prefetched regs into place
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int ii = 0; ii < 2; ++ii)
          afr[ii][ks] = anx[ii][ks];
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
  const size_t lds_bytes = 10 * HALF_ELEMS * sizeof(short);  // 160 KiB
  hipError_t se = hipFuncSetAttribute(
      (const void*)&gemm_v11_k,
      hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds_bytes);
  if (se != hipSuccess) {
    printf("LDS 160KiB not grantable: %s\n", hipGetErrorString(se));
    return 1;
  }
  auto launch = [&]() {
    hipLaunchKernelGGL(gemm_v11_k, dim3(grid), dim3(THREADS), lds_bytes,
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
      hipError_t le = hipDeviceSynchronize();
      if (le != hipSuccess) {
        printf("launch/sync failed: %s\n", hipGetErrorString(le));
        return 1;
      }
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
    printf("refcheck %dx%dx%d v11 x%d: %s (%d bad)\n", M, N, K, repeats,
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
    printf("perf %dx%dx%d v11: %.3f ms/iter, %.0f TFLOP/s\n", M, N, K,
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
    bad += run_case(256, 256, 64, true, 1, 3);
    bad += run_case(256, 256, 128, true, 1, 3);
    bad += run_case(512, 512, 192, true, 1, 3);
    bad += run_case(512, 512, 768, true, 1, 5);
    if (bad) {
      printf("REFCHECK FAILED - do not trust perf numbers\n");
      return 1;
    }
  }
  run_case(4096, 4096, 4096, false, 10);
  run_case(8192, 8192, 8192, false, 5);
  run_case(32768, 3072, 768, false, 10);
  run_case(32768, 768, 3072, false, 10);
  run_case(32768, 768, 768, false, 10);
  run_case(32768, 2304, 768, false, 10);
  return 0;
}
