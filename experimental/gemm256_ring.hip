// EXPERIMENTAL — standalone probe for the round-2 GEMM schedule.
//
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 \
//       experimental/gemm256_ring.hip -o /tmp/gemm256r && /tmp/gemm256r
//
// C[M,N] = A[M,K] @ W[N,K]^T, bf16 in/out, fp32 accum.
//
// 256x256 tile, BK=64, 512 threads = 8 waves (2M x 4N), per-wave output
// 128x64. Schedule: 4 phases per K-tile, each phase
//   { ds_read frags | issue 1 half-tile global_load_lds | barrier |
//     lgkmcnt(0) | setprio(1) 16 MFMA setprio(0) | [vmcnt] | barrier }
// with a COUNTED vmcnt once per K-tile (phase 3) — never 0 in steady
// state.
//
// LDS: two 4-deep rings of 128x64 bf16 half-tiles (A ring + B ring),
// 8 x 16 KiB = 128 KiB. Half h of K-tile t lives in ring slot (2t+h)&3.
//
// Staging schedule (during tile u's 4 phases):
//   phase 0: A-half0(u+1)   phase 1: A-half1(u+1)
//   phase 2: B-half0(u+2)   phase 3: B-half1(u+2)
// B halves are read only at phase 0 of their tile (frags held in
// registers all 4 phases), so their ring slot frees after phase 0 and
// they can be staged nearly two tiles ahead; A-half wr is read by the
// wr-waves at every phase, freeing only at the tile boundary.
//
// WAIT DERIVATION (cross-wave safety). vmcnt counts only the waiting
// wave's own loads, but every wave reads LDS segments written by the
// other waves' global_load_lds. A staged half-tile is therefore only
// safe to read at phase p if EVERY wave executed a vmcnt wait covering
// its own 2 staging calls for that half-tile BEFORE a barrier that
// precedes phase p. With the end-of-phase barrier as the separator:
//   needed by end of tile u phase 3 (reads start at u+1 phase 0):
//     A(u+1) halves  — staged at u phases 0,1
//     B(u+1) halves  — staged at u-1 phases 2,3 (older, covered)
//   at the phase-3 wait the per-thread issue queue (newest first) is
//     [B1(u+2), B0(u+2), A1(u+1), A0(u+1), ...]
//   allowing the 2 newest stagings (2 calls each) to stay in flight:
//     s_waitcnt vmcnt(4)        (vmcnt(0) for the last two tiles)
// So A halves get ~3 MFMA phases of landing slack, B halves 4-5, and
// the only wait is one counted vmcnt per K-tile.
//
// WAR safety: ds_reads of phase p complete before that phase's MFMAs
// (lgkmcnt(0)), which precede the end-of-phase barrier; every staging
// call that reuses a ring slot is issued at least one full phase after
// the slot's last read phase (A: slot of (h,u-1) reused at u phase h,
// last read u-1 phase 3; B: slot of (h,u) reused at u phase 2+h, last
// read u phase 0).
//
// Design refs: /opt/skills/guides/cdna_hip_programming.md §5 (8-phase
// template, T1 XCD swizzle, T2 st_16x32 swizzle via pre-swizzled
// global source, T3+T4 counted vmcnt, T5 setprio).

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
constexpr int THREADS = 512;  // 8 waves: 2 (M) x 4 (N)
constexpr int HALF_ELEMS = 128 * BK;  // one 128x64 half-tile

__device__ __forceinline__ short f2bf(float f) {
  union { float f; unsigned u; } c;
  c.f = f;
  unsigned lsb = (c.u >> 16) & 1u;
  c.u += 0x7fffu + lsb;
  return (short)(c.u >> 16);
}

// XOR-swizzle involution on a [row][64] bf16 half-tile byte offset
// (rows are 128 B): XOR byte bits 4-6 with row bits 0-2 (byte bits
// 7-9). 16 consecutive rows at one 16 B column spread over 8 distinct
// 16 B slots -> 2 lanes/bank (free on CDNA4). 16 B chunks stay intact
// and the map is its own inverse.
__device__ __forceinline__ int swz(int byte_off) {
  return byte_off ^ (((byte_off >> 7) & 7) << 4);
}

template <bool SWZ>
__device__ __forceinline__ bf16x8 ld_frag(const short* slot, int row,
                                          int kk) {
  int byte = (row * BK + kk) * 2;
  if (SWZ) byte = swz(byte);
  return *(const bf16x8*)((const char*)slot + byte);
}

template <bool SWZ>
__global__ __launch_bounds__(THREADS) void gemm256_ring_k(
    const short* __restrict__ A, const short* __restrict__ W,
    short* __restrict__ C, int M, int N, int K) {
  extern __shared__ short lds[];

  // T1: bijective XCD-aware remap (8 XCDs, private L2 per XCD)
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
  const int wr = wave >> 2;  // 0..1 -> output rows [wr*128, +128)
  const int wc = wave & 3;   // 0..3 -> output cols [wc*64, +64)
  const int frag_row = lane % 16;
  const int frag_k = (lane / 16) * 8;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // ring slots: A slots 0..3, B slots 4..7, each 128x64 bf16 (16 KiB)
  auto slotA = [&](int t, int h) {
    return lds + (size_t)((2 * t + h) & 3) * HALF_ELEMS;
  };
  auto slotB = [&](int t, int h) {
    return lds + (size_t)(4 + ((2 * t + h) & 3)) * HALF_ELEMS;
  };

  // Stage half-tile h (rows [h*128, h*128+128)) of operand g's K-tile t
  // into its ring slot: 2 cooperative global_load_lds calls, 512
  // threads x 16 B each. The LDS write is linear (wave-uniform base +
  // lane*16); with SWZ the per-lane GLOBAL source is pre-permuted with
  // the same involution the ds_reads apply.
  auto stage_half = [&](short* slot, const short* g, long long row0,
                        int h, int k0, int ld) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      const int e_lin = (c * THREADS + tid) * 8;  // element in half-tile
      int e = e_lin;
      if (SWZ) e = swz(e_lin * 2) / 2;
      const int row = e / BK, kk = e % BK;
      const short* gp =
          g + (row0 + h * 128 + row) * (long long)ld + k0 + kk;
      short* lp = slot + (c * THREADS + (tid & ~63)) * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp,
          (__attribute__((address_space(3))) unsigned int*)lp, 16, 0, 0);
    }
  };

  const int T = K / BK;  // number of K-tiles

  // Prologue: A+B of tile 0, B of tile 1 (the loop stages A(u+1) and
  // B(u+2) during tile u). Full drain once.
  stage_half(slotA(0, 0), A, a_row0, 0, 0, K);
  stage_half(slotA(0, 1), A, a_row0, 1, 0, K);
  stage_half(slotB(0, 0), W, b_row0, 0, 0, K);
  stage_half(slotB(0, 1), W, b_row0, 1, 0, K);
  if (T > 1) {
    stage_half(slotB(1, 0), W, b_row0, 0, BK, K);
    stage_half(slotB(1, 1), W, b_row0, 1, BK, K);
  }
  __builtin_amdgcn_s_waitcnt(0);  // vmcnt(0) lgkmcnt(0)
  __syncthreads();

  for (int u = 0; u < T; ++u) {
    const short* sA = slotA(u, wr);
    bf16x8 bfr[4][2];
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      // ds_read this phase's fragments: 2 A-frags x 2 K-steps, plus
      // the whole B panel (4 frags x 2 K-steps) at phase 0.
      bf16x8 afr[2][2];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int ii = 0; ii < 2; ++ii)
          afr[ii][ks] = ld_frag<SWZ>(sA, q * 32 + ii * 16 + frag_row,
                                     ks * 32 + frag_k);
      if (q == 0) {
        const short* sB = slotB(u, wc >> 1);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            bfr[j][ks] = ld_frag<SWZ>(
                sB, (wc & 1) * 64 + j * 16 + frag_row, ks * 32 + frag_k);
      }

      // issue this phase's half-tile prefetch
      if (q == 0 && u + 1 < T)
        stage_half(slotA(u + 1, 0), A, a_row0, 0, (u + 1) * BK, K);
      else if (q == 1 && u + 1 < T)
        stage_half(slotA(u + 1, 1), A, a_row0, 1, (u + 1) * BK, K);
      else if (q == 2 && u + 2 < T)
        stage_half(slotB(u + 2, 0), W, b_row0, 0, (u + 2) * BK, K);
      else if (q == 3 && u + 2 < T)
        stage_half(slotB(u + 2, 1), W, b_row0, 1, (u + 2) * BK, K);

      __builtin_amdgcn_s_barrier();
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
        // counted wait (see header): everything up to A1(u+1) must
        // have landed before the barrier below; the B(u+2) halves
        // (2 calls each) may stay in flight.
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

// ---------------------------------------------------------------------
// Baseline for A/B: the validated 2-phase drain-per-K-tile double
// buffer at the same 256x256 geometry (from gemm256_dbuf.hip).
// ---------------------------------------------------------------------

template <bool SWZ>
__global__ __launch_bounds__(THREADS) void gemm256_drain_k(
    const short* __restrict__ A, const short* __restrict__ W,
    short* __restrict__ C, int M, int N, int K) {
  extern __shared__ short lds[];

  const int nwg = gridDim.x;
  const int q = nwg / 8, r = nwg % 8;
  const int xcd = blockIdx.x % 8, idx = blockIdx.x / 8;
  const int wgid =
      (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
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

  auto stage = [&](short* ldst, const short* g, long long row0, int k0,
                   int ld) {
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const int e_base = (s * THREADS + tid) * 8;
      int e = e_base;
      if (SWZ) e = swz(e_base * 2) / 2;
      const int row = e / BK, kk = e % BK;
      const short* gp = g + (row0 + row) * (long long)ld + k0 + kk;
      short* lp = ldst + ((s * THREADS + (tid & ~63)) * 8);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp,
          (__attribute__((address_space(3))) unsigned int*)lp, 16, 0, 0);
    }
  };

  auto bufA = [&](int b) { return lds + (size_t)b * 2 * BM * BK; };
  auto bufB = [&](int b) { return lds + ((size_t)b * 2 + 1) * BM * BK; };

  stage(bufA(0), A, a_row0, 0, K);
  stage(bufB(0), W, b_row0, 0, K);
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();

  int cur = 0;
  const int ntiles = K / BK;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {
      stage(bufA(cur ^ 1), A, a_row0, (t + 1) * BK, K);
      stage(bufB(cur ^ 1), W, b_row0, (t + 1) * BK, K);
    }
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a[8], b[4];
#pragma unroll
      for (int i = 0; i < 8; ++i)
        a[i] = ld_frag<SWZ>(bufA(cur), wr * 128 + i * 16 + frag_row,
                            ks * 32 + frag_k);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        b[j] = ld_frag<SWZ>(bufB(cur), wc * 64 + j * 16 + frag_row,
                            ks * 32 + frag_k);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    __builtin_amdgcn_s_waitcnt(0);
    __syncthreads();
    cur ^= 1;
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

// ---------------------------------------------------------------------
// Harness: repeated refcheck (race screen) + perf on BERT shapes,
// 4096^3, 8192^3.
// ---------------------------------------------------------------------

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

template <bool SWZ, bool RING>
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
  const size_t lds_bytes = 2 * 2 * (size_t)BM * BK * sizeof(short);
  const void* kfn = RING ? (const void*)&gemm256_ring_k<SWZ>
                         : (const void*)&gemm256_drain_k<SWZ>;
  HIP_CHECK(hipFuncSetAttribute(
      kfn, hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds_bytes));
  auto launch = [&]() {
    if (RING)
      hipLaunchKernelGGL((gemm256_ring_k<SWZ>), dim3(grid), dim3(THREADS),
                         lds_bytes, 0, dA, dW, dC, M, N, K);
    else
      hipLaunchKernelGGL((gemm256_drain_k<SWZ>), dim3(grid), dim3(THREADS),
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
    printf("refcheck %dx%dx%d swz=%d ring=%d x%d: %s (%d bad)\n", M, N,
           K, (int)SWZ, (int)RING, repeats, bad ? "FAIL" : "ok", bad);
  } else {
    launch();  // warm
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
    printf("perf %dx%dx%d swz=%d ring=%d: %.3f ms/iter, %.0f TFLOP/s\n",
           M, N, K, (int)SWZ, (int)RING, ms / iters, tf);
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
    // race screen: repeated refcheck, K small and large tile counts
    bad += run_case<false, true>(256, 256, 64, true, 1, 3);
    bad += run_case<false, true>(256, 256, 128, true, 1, 3);
    bad += run_case<false, true>(512, 512, 192, true, 1, 3);
    bad += run_case<false, true>(512, 512, 768, true, 1, 5);
    bad += run_case<true, true>(256, 256, 64, true, 1, 3);
    bad += run_case<true, true>(256, 256, 128, true, 1, 3);
    bad += run_case<true, true>(512, 512, 192, true, 1, 3);
    bad += run_case<true, true>(512, 512, 768, true, 1, 5);
    bad += run_case<true, false>(512, 512, 768, true, 1, 2);
    if (bad) {
      printf("REFCHECK FAILED - do not trust perf numbers\n");
      return 1;
    }
  }
  // A/B perf: ring vs drain, swizzle on/off
  run_case<false, false>(4096, 4096, 4096, false, 10);
  run_case<true, false>(4096, 4096, 4096, false, 10);
  run_case<false, true>(4096, 4096, 4096, false, 10);
  run_case<true, true>(4096, 4096, 4096, false, 10);
  run_case<true, true>(8192, 8192, 8192, false, 5);
  run_case<true, false>(8192, 8192, 8192, false, 5);
  // BERT-base seq512 bs64 shapes (M=32768): FFN up (N=3072 K=768),
  // FFN down (N=768 K=3072), QKV-ish proj (N=768 K=768, N=2304 K=768)
  run_case<true, true>(32768, 3072, 768, false, 10);
  run_case<true, false>(32768, 3072, 768, false, 10);
  run_case<true, true>(32768, 768, 3072, false, 10);
  run_case<true, true>(32768, 768, 768, false, 10);
  run_case<true, true>(32768, 2304, 768, false, 10);
  return 0;
}
