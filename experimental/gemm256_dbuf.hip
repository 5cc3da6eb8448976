// EXPERIMENTAL — not built into sparkdl._C; standalone probe for the
// next GEMM schedule (NOTES-round2.md §1). Compile + run on an MI355X:
//
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 \
//       experimental/gemm256_dbuf.hip -o /tmp/gemm256 && /tmp/gemm256
//
// C[M,N] = A[M,K] @ W[N,K]^T, bf16 in, bf16 out, fp32 accum.
//
// Structure ("minimum 2-phase" deep-buffer pattern, CDNA4 guide §5.5 T3
// recipe): 256x256 tile, K-step 64, 512 threads = 8 waves (2M x 4N),
// per-wave output 128x64 = 8x4 fragments x 2 K-steps = 64 MFMA/K-tile.
// Double-buffered LDS (2 x (A 256x64 + B 256x64) bf16 = 128 KiB):
//   prologue: stage buf0; vmcnt(0); barrier
//   loop:     stage buf^1 for t+1 ; ds_read + 64 MFMA from buf ;
//             vmcnt(0) ; barrier ; swap
// plus the LDS st_16x32 XOR swizzle on the K-major tiles: ds_read_b128
// of a [row][64] bf16 tile at stride 128 B is otherwise a 16-way bank
// conflict. global_load_lds writes LINEARLY (wave-uniform base +
// lane*16), so the per-lane GLOBAL source is pre-permuted with the same
// involution the reads apply (both-sides-or-neither rule).
//
// Round-1 probe result (run on MI355X): the first version staged only
// half of each tile (2 staging calls instead of 4) and failed refcheck
// for columns/rows >= 128, exactly as that bug predicts; fixed below,
// NOT yet re-validated on hardware — round 2 starts by re-running this
// harness (compile+refcheck+perf is a ~40 s GPU call, no torch import).
//
// Round-2 work after validation (expected +30-40%): split the K-step
// into the 8-phase interleave with counted vmcnt (never 0 in the loop)
// + s_setprio around the MFMA clusters.

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
      printf("HIP error %s at %s:%d\n", hipGetErrorString(e), __FILE__, \
             __LINE__);                                                  \
      exit(1);                                                           \
    }                                                                    \
  } while (0)

__device__ __forceinline__ float bf2f(short s) {
  union { float f; unsigned u; } c;
  c.u = ((unsigned)(unsigned short)s) << 16;
  return c.f;
}
__device__ __forceinline__ short f2bf(float f) {
  union { float f; unsigned u; } c;
  c.f = f;
  unsigned lsb = (c.u >> 16) & 1u;
  c.u += 0x7fffu + lsb;
  return (short)(c.u >> 16);
}

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int THREADS = 512;  // 8 waves: 2 (M) x 4 (N)

// XOR-swizzle involution on a [row][BK=64] bf16 tile byte offset
// (rows are 128 B): XOR byte bits 4-6 with row bits 0-2 (= byte bits
// 7-9). 16 consecutive rows at one 16 B column spread over 8 distinct
// 16 B slots -> 2 lanes/bank, which is free on CDNA4. Keeps 16 B
// chunks intact (bits 0-3 untouched) and is its own inverse (bits 7-9
// are not modified).
__device__ __forceinline__ int swz(int byte_off) {
  return byte_off ^ (((byte_off >> 7) & 7) << 4);
}

template <bool SWZ>
__device__ __forceinline__ bf16x8 ld_frag_g(const short* ldst, int row,
                                            int kk) {
  int byte = (row * BK + kk) * 2;
  if (SWZ) byte = swz(byte);
  return *(const bf16x8*)((const char*)ldst + byte);
}

template <bool SWZ>
__global__ __launch_bounds__(THREADS) void gemm256_k(
    const short* __restrict__ A, const short* __restrict__ W,
    short* __restrict__ C, int M, int N, int K) {
  // 2 buffers x (A-tile + B-tile), each tile 256x64 bf16 = 32 KiB;
  // 128 KiB total -> dynamic LDS (static __shared__ caps at 64 KiB)
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
  const int wr = wave >> 2;  // 0..1 -> rows [wr*128, +128)
  const int wc = wave & 3;   // 0..3 -> cols [wc*64, +64)

  const int frag_row = lane % 16;
  const int frag_k = (lane / 16) * 8;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // Cooperative staging of one 256x64 operand tile: 512 threads x
  // 4 calls x 16 B = 32 KiB. Linear LDS element index e = t*8 within
  // the tile maps to (row = e/BK, k = e%BK); when SWZ, the global
  // source is pre-permuted so that a swizzled ds_read sees the right
  // data in the linearly-written LDS.
  auto stage = [&](short* ldst, const short* g, long long row0, int k0,
                   int ld) {
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const int e_base = (s * THREADS + tid) * 8;
      int e = e_base;
      if (SWZ) {
        // the byte the linear write at e_base*2 will OCCUPY is read
        // back as swz(e_base*2); feed it the element that belongs there
        e = swz(e_base * 2) / 2;
      }
      const int row = e / BK, kk = e % BK;
      const short* gp = g + (row0 + row) * (long long)ld + k0 + kk;
      short* lp = ldst + ((s * THREADS + (tid & ~63)) * 8);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp,
          (__attribute__((address_space(3))) unsigned int*)lp, 16, 0, 0);
    }
  };

  auto ld_frag = [&](const short* ldst, int row, int kk) -> bf16x8 {
    int byte = (row * BK + kk) * 2;
    if (SWZ) byte = swz(byte);
    return *(const bf16x8*)((const char*)ldst + byte);
  };

  // buffer b: A at lds + b*2*BM*BK, B at lds + (b*2+1)*BM*BK
  auto bufA = [&](int b) { return lds + (size_t)b * 2 * BM * BK; };
  auto bufB = [&](int b) { return lds + ((size_t)b * 2 + 1) * BM * BK; };

  // prologue: stage K-tile 0 into buf 0
  stage(bufA(0), A, a_row0, 0, K);
  stage(bufB(0), W, b_row0, 0, K);
  __builtin_amdgcn_s_waitcnt(0 /* vmcnt(0) lgkmcnt(0) */);
  __syncthreads();

  int cur = 0;
  const int ntiles = K / BK;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {  // prefetch next K-tile into the other buffer
      stage(bufA(cur ^ 1), A, a_row0, (t + 1) * BK, K);
      stage(bufB(cur ^ 1), W, b_row0, (t + 1) * BK, K);
    }
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {  // two 16x16x32 K-steps per tile
      bf16x8 a[8], b[4];
#pragma unroll
      for (int i = 0; i < 8; ++i)
        a[i] = ld_frag(bufA(cur), wr * 128 + i * 16 + frag_row,
                       ks * 32 + frag_k);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        b[j] = ld_frag(bufB(cur), wc * 64 + j * 16 + frag_row,
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
  for (int i = 0; i < 8; ++i) {
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
  }
}

// ---------------------------------------------------------------------
// Variant 2: 4-phase counted-vmcnt pipeline (the T3+T4 structure).
//
// Same geometry; each K-tile is computed in 4 phases (phase q = 2 of
// the 8 A-row fragments x all 4 B fragments x 2 K-steps = 16 MFMA).
// Each phase also issues 2 of the NEXT K-tile's 8 cooperative staging
// calls (order: B-half0, B-half1, A-half0, A-half1), and waits with a
// COUNTED vmcnt — never 0 in the main loop — sized so exactly the
// staging calls this wave is about to read have landed:
//
//   per-thread call completion is in issue order (vmcnt semantics);
//   at phase q (after issuing 2 calls), outstanding =
//       (8 - completed-of-current-tile) + 2(q+1)
//   the wave needs current-tile calls <= Cmax(q) done, where
//       Cmax = 4 + 2*wr + (q>=2)   (A calls dominate; B calls are 0-3)
//   -> s_waitcnt vmcnt((7 - Cmax) + 2(q+1))    [wave-uniform branch]
//
// One s_barrier per phase (raw, no vmcnt(0) drain — that drain is the
// structural stall of the 2-phase version) + s_setprio around the MFMA
// cluster. Buffers: staging for tile t+1 writes buf^1 while buf is
// read, so two buffers suffice and the only cross-wave hazard is the
// tile-boundary barrier.
// ---------------------------------------------------------------------

template <bool SWZ>
__global__ __launch_bounds__(THREADS) void gemm256_8ph_k(
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

  auto bufA = [&](int b) { return lds + (size_t)b * 2 * BM * BK; };
  auto bufB = [&](int b) { return lds + ((size_t)b * 2 + 1) * BM * BK; };

  // one cooperative 16-B staging call: piece c of 8 for one K-tile,
  // order [B0 B1 A0 A1] x 2 calls each; piece c covers tile elements
  // [c*4096, (c+1)*4096) of the B (c<4) or A (c>=4) operand tile.
  auto stage_piece = [&](int buf, int c, int k0) {
    const bool isB = c < 4;
    const int cc = isB ? c : c - 4;
    short* ldst = (isB ? bufB(buf) : bufA(buf)) + cc * 4096;
    const short* g = isB ? W : A;
    const long long row0 = isB ? b_row0 : a_row0;
    const int e_base0 = cc * 4096;  // element offset within the tile
    const int e_base = e_base0 + tid * 8;
    int e = e_base;
    if (SWZ) e = swz(e_base * 2) / 2;
    const int row = e / BK, kk = e % BK;
    const short* gp = g + (row0 + row) * (long long)K + k0 + kk;
    // wave-uniform LDS base: this wave's 64 lanes write 1 KiB linearly
    short* lp = ldst + (tid & ~63) * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)gp,
        (__attribute__((address_space(3))) unsigned int*)lp, 16, 0, 0);
  };

  // prologue: stage all 8 pieces of K-tile 0 into buf 0, drain, barrier
  for (int c = 0; c < 8; ++c) stage_piece(0, c, 0);
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();

  const int ntiles = K / BK;
  int cur = 0;
  for (int t = 0; t < ntiles; ++t) {
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      if (t + 1 < ntiles) {  // issue 2 of the next tile's 8 pieces
        stage_piece(cur ^ 1, 2 * q, (t + 1) * BK);
        stage_piece(cur ^ 1, 2 * q + 1, (t + 1) * BK);
        // counted wait: current tile's needed pieces have landed
        if (t == 0) {
          // first tile was fully drained in the prologue; only pace
          // the new issues loosely
          asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
        } else if (wr == 0) {
          if (q == 0) asm volatile("s_waitcnt vmcnt(5)" ::: "memory");
          else if (q == 1) asm volatile("s_waitcnt vmcnt(7)" ::: "memory");
          else if (q == 2) asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
          // q == 3: everything needed already covered
        } else {
          if (q == 0) asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
          else if (q == 1) asm volatile("s_waitcnt vmcnt(5)" ::: "memory");
          else if (q == 2) asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
          else asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
        }
      } else {
        // last tile: no new issues; make sure the pieces are in
        if (q == 0) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }

      bf16x8 a[2][2], b[4][2];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
        for (int ii = 0; ii < 2; ++ii)
          a[ii][ks] = ld_frag_g<SWZ>(
              bufA(cur), wr * 128 + (q * 2 + ii) * 16 + frag_row,
              ks * 32 + frag_k);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          b[j][ks] = ld_frag_g<SWZ>(bufB(cur),
                                    wc * 64 + j * 16 + frag_row,
                                    ks * 32 + frag_k);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int ii = 0; ii < 2; ++ii)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[q * 2 + ii][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a[ii][ks], b[j][ks], acc[q * 2 + ii][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      asm volatile("" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
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
// Harness: refcheck at small sizes, TFLOPs at 4096^3 and 8192^3.
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

template <bool SWZ, bool PIPE = false>
static int run_case(int M, int N, int K, bool check, int iters) {
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
  const void* kfn = PIPE ? (const void*)&gemm256_8ph_k<SWZ>
                         : (const void*)&gemm256_k<SWZ>;
  HIP_CHECK(hipFuncSetAttribute(
      kfn, hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds_bytes));
  auto launch = [&]() {
    if (PIPE)
      hipLaunchKernelGGL((gemm256_8ph_k<SWZ>), dim3(grid), dim3(THREADS),
                         lds_bytes, 0, dA, dW, dC, M, N, K);
    else
      hipLaunchKernelGGL((gemm256_k<SWZ>), dim3(grid), dim3(THREADS),
                         lds_bytes, 0, dA, dW, dC, M, N, K);
  };
  launch();
  HIP_CHECK(hipDeviceSynchronize());

  int bad = 0;
  if (check) {
    std::vector<float> ref((size_t)M * N);
    cpu_ref(hA, hW, ref, M, N, K);
    std::vector<short> hC((size_t)M * N);
    HIP_CHECK(hipMemcpy(hC.data(), dC, hC.size() * 2,
                        hipMemcpyDeviceToHost));
    for (size_t i = 0; i < hC.size(); ++i) {
      union { float f; unsigned u; } c;
      c.u = ((unsigned)(unsigned short)hC[i]) << 16;
      const float got = c.f, want = ref[i];
      if (fabsf(got - want) > 2e-2f + 2e-2f * fabsf(want)) {
        if (bad < 5)
          printf("  mismatch [%zu]: got %f want %f\n", i, got, want);
        ++bad;
      }
    }
    printf("refcheck %dx%dx%d swz=%d pipe=%d: %s (%d bad)\n", M, N, K,
           (int)SWZ, (int)PIPE, bad ? "FAIL" : "ok", bad);
  } else {
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
    printf("perf %dx%dx%d swz=%d pipe=%d: %.3f ms/iter, %.0f TFLOP/s\n",
           M, N, K, (int)SWZ, (int)PIPE, ms / iters, tf);
  }
  HIP_CHECK(hipFree(dA));
  HIP_CHECK(hipFree(dW));
  HIP_CHECK(hipFree(dC));
  return bad;
}

int main() {
  int bad = 0;
  bad += run_case<false>(256, 256, 64, true, 1);
  bad += run_case<false>(512, 512, 192, true, 1);
  bad += run_case<true>(256, 256, 64, true, 1);
  bad += run_case<true>(512, 512, 192, true, 1);
  bad += run_case<false, true>(256, 256, 64, true, 1);
  bad += run_case<false, true>(512, 512, 192, true, 1);
  bad += run_case<true, true>(256, 256, 64, true, 1);
  bad += run_case<true, true>(512, 512, 192, true, 1);
  if (bad) {
    printf("REFCHECK FAILED - do not trust perf numbers\n");
    return 1;
  }
  run_case<false>(4096, 4096, 4096, false, 10);
  run_case<true>(4096, 4096, 4096, false, 10);
  run_case<false, true>(4096, 4096, 4096, false, 10);
  run_case<true, true>(4096, 4096, 4096, false, 10);
  run_case<true, true>(8192, 8192, 8192, false, 5);
  run_case<true>(32768, 3072, 768, false, 10);
  run_case<true, true>(32768, 3072, 768, false, 10);
  return 0;
}
