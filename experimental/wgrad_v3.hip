// EXPERIMENTAL — split-M wgrad GEMM: dW[N,K] = dZ^T[N,M] @ X[M,K].
//
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 \
//       experimental/wgrad_v3.hip -o /tmp/wg && /tmp/wg
//
// MEASURED (MI355X): refcheck-clean; 506-534 TF on the big BERT wgrad
// shapes — 1.9x over v2's row-major tr reads (272, 4-way conflicted)
// and 1.5x over v1's scalar transpose scatter (336). A 3-deep counted
// ring was neutral (the drain was not the bound). Tensile's wgrad
// kernels reach ~650 TF on the same shapes, so the library keeps the
// production wgrad slot for now; remaining gap candidates: tr-read
// issue rate (24 reads : 16 MFMA per m-tile per wave) — a larger
// K-tile (TM=128) or fused dual-subtile reads would amortize.
//
// v3 = v2 + conflict-free BLOCKED LDS layout: the tile is stored as
// [m/4][minor/16][4][16] so every [4][16] transpose subtile is 128
// CONTIGUOUS bytes — the 16-lane tr gather then touches consecutive
// banks (v2's row-major layout put the subtile's 4 rows 512 B apart:
// same banks -> the measured 4-way conflict bound at 272 TF).
// Staging stays pure global_load_lds: the per-lane GLOBAL source is
// pre-permuted so the linear LDS write produces the blocked layout
// (16-byte chunks stay intact: a chunk is 8 consecutive minor cols of
// one m row in both layouts).
//
// v2: tiles staged LINEARLY ([m][minor]) via global_load_lds; the
// MFMA fragments are read with ds_read_b64_tr_b16 hardware transpose
// reads, whose lane semantics were pinned by tr_b16_probe.hip: per
// 16-lane group a [4 rows][16 cols] subtile transposes so lane l
// receives column l&15 with elems = the 4 rows; each lane's address
// independently selects its source b64, so strided subtiles of the
// [64 m][minor] tile work directly. This replaces v1's scalar
// ds_write transpose scatter (its measured bound: 336 TF).
//
// MEASURED: refcheck-clean; 272-278 TF on the BERT wgrad shapes —
// BELOW v1's 336 (and batching the reads behind one lgkmcnt changed
// nothing), consistent with the guide's m217 finding that tr reads
// from simply-subtiled layouts carry ~4-way bank conflicts that
// address swizzles cannot fix; the conflict-free 8x[32][16] LDS
// subtiling (s21) is the round-3 follow-up. Two asm lessons captured:
// early-clobber ("=&v") is mandatory on multi-read asm blocks (the
// first read clobbered the second's address register -> silent data
// corruption caught by refcheck), and integer self-identifying test
// values must stay < 256 to be bf16-exact (the probe's apparent
// "group 3 corruption" was bf16 rounding of 300-1100).
//
// Geometry: output tile 256(N) x 128(K), 16 waves (4x4), per-wave
// 64x32 (acc 4x2); m-tiles of 64 (2 K-steps); grid =
// (N/256) x (K/128) x SPLITS with fp32 atomicAdd partials into dW.
//
// MEASURED (MI355X): 336/340/298/201 TF on the four BERT wgrad shapes
// with the wide swizzle (the narrow row-bits-0-2 swizzle left the
// transpose scatter 32-way conflicted: 98 TF). A 256x256 tile variant
// measured SLOWER (304 TF + spills): the bound is the scalar ds_write
// transpose staging, not MFMA count. Tensile's wgrad kernels reach
// ~650 TF on these shapes, so the library keeps the wgrad slot until
// a ds_read_b64_tr_b16 fragment path replaces the scatter (round-3
// queue item 2 in NOTES-round2.md).
// Tensile's wgrad kernels on the BERT shapes (bs64 s512) measure
// 149-237 us — the bar.
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

constexpr int TN = 256;   // dW rows per tile (dZ columns)
constexpr int TK = 128;   // dW cols per tile (X columns)
constexpr int TM = 64;    // contraction chunk
constexpr int THREADS = 1024;

__device__ __forceinline__ float bf2f(short s) {
  union { float f; unsigned u; } c;
  c.u = ((unsigned)(unsigned short)s) << 16;
  return c.f;
}
typedef short b16x4 __attribute__((ext_vector_type(4)));

// blocked layout: element (m, c) of a [64 m][W c] tile lives at
//   blk_off(m, c, W) = (((m >> 2) * (W >> 4) + (c >> 4)) << 6)
//                      + ((m & 3) << 4) + (c & 15)
__device__ __forceinline__ int blk_off(int m, int c, int W) {
  return (((m >> 2) * (W >> 4) + (c >> 4)) << 6) + ((m & 3) << 4) +
         (c & 15);
}

// Transpose fragment read: returns the MFMA operand fragment
// [row = minor-col l%16 of block col0][m = mstep + (l/16)*8 .. +8]
// from a LINEAR [64 m][stride] bf16 tile, via two hardware transpose
// reads of [4][16] subtiles (semantics: tr_b16_probe.hip).
__device__ __forceinline__ bf16x8 tr_frag(const short* tile, int lane,
                                          int stride, int mstep,
                                          int col0) {
  const unsigned base = (unsigned)(unsigned long long)(
      (const __attribute__((address_space(3))) short*)tile);
  const int lg = lane & 15;
  const int m0 = mstep + (lane >> 4) * 8;
  // subtile (m0, col0) is contiguous in the blocked layout; lane l
  // supplies its row lg>>2, cols 4*(lg&3)..+4
  const unsigned a0 =
      base + (blk_off(m0, col0, stride) + (lg >> 2) * 16 +
              4 * (lg & 3)) * 2;
  b16x4 v0, v1;
  // issue-only: the caller waits lgkmcnt(0) once per batch and pins
  // each fragment with lds_pin() before use
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %3"
      : "=&v"(v0), "=&v"(v1)  // early-clobber: inputs must not share
      : "v"(a0), "v"(a0 + (unsigned)(stride >> 4) * 64u * 2u)
      : "memory");
  bf16x8 r;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    r[j] = v0[j];
    r[4 + j] = v1[j];
  }
  return r;
}

__device__ __forceinline__ void lds_wait_all() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
}
__device__ __forceinline__ void lds_pin(bf16x8& v) {
  asm volatile("" : "+v"(v));
}

__global__ __launch_bounds__(THREADS) void wgrad_k(
    const short* __restrict__ dZ, const short* __restrict__ X,
    float* __restrict__ dW, long long M, int N, int K, int splits) {
  // LDS: dzT [256 n][64 m] (32 KiB) + xT [128 k][64 m] (16 KiB), dbuf
  __shared__ short ldz[3][TN * TM];
  __shared__ short lx[3][TK * TM];

  const int ntn = N / TN, ntk = K / TK;
  const int tile = blockIdx.x % (ntn * ntk);
  const int split = blockIdx.x / (ntn * ntk);
  const int n0 = (tile / ntk) * TN;
  const int k0 = (tile % ntk) * TK;

  const long long mtiles_total = M / TM;
  const long long per = (mtiles_total + splits - 1) / splits;
  const long long mt0 = split * per;
  const long long mt1 = (mt0 + per < mtiles_total) ? mt0 + per
                                                   : mtiles_total;
  if (mt0 >= mt1) return;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wn = wave >> 2;  // 0..3: dW rows [wn*64, +64)
  const int wk = wave & 3;   // 0..3: dW cols [wk*32, +32)
  const int frag_row = lane % 16;
  const int frag_k = (lane / 16) * 8;

  f32x4 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // linear staging: LDS tiles are [64 m][minor], matching global rows
  auto stage = [&](int buf, long long m0) {
    // linear LDS position p (in 8-short chunks) holds blocked-layout
    // element: invert blk_off for chunk starts (c&15 in {0,8}):
    //   p8 = p >> 4 grid: blkid = p/64, within = p%64
    //   m = (blkid / (W/16))*4 + (within>>4); c = (blkid % (W/16))*16
    //       + (within & 15)
    // dZ tile: 64 m x 256 n -> 2 sweeps
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      const int p = (s * THREADS + tid) * 8;
      const int blkid = p >> 6, within = p & 63;
      const int m = (blkid / (TN >> 4)) * 4 + (within >> 4);
      const int c = (blkid % (TN >> 4)) * 16 + (within & 15);
      const short* gp = dZ + (m0 + m) * (long long)N + n0 + c;
      short* lp = ldz[buf] + ((s * THREADS + (tid & ~63)) * 8);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp,
          (__attribute__((address_space(3))) unsigned int*)lp, 16, 0, 0);
    }
    // X tile: 64 m x 128 k -> 1 sweep
    {
      const int p = tid * 8;
      const int blkid = p >> 6, within = p & 63;
      const int m = (blkid / (TK >> 4)) * 4 + (within >> 4);
      const int c = (blkid % (TK >> 4)) * 16 + (within & 15);
      const short* gp = X + (m0 + m) * (long long)K + k0 + c;
      short* lp = lx[buf] + ((tid & ~63) * 8);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp,
          (__attribute__((address_space(3))) unsigned int*)lp, 16, 0, 0);
    }
  };

  stage(0, mt0 * TM);
  if (mt0 + 1 < mt1) stage(1, (mt0 + 1) * TM);
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();

  for (long long mt = mt0; mt < mt1; ++mt) {
    const int cur = (int)((mt - mt0) % 3);
    if (mt + 2 < mt1) stage((cur + 2) % 3, (mt + 2) * TM);
    bf16x8 a[2][4], b[2][2];
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
      for (int i = 0; i < 4; ++i)
        a[ks][i] = tr_frag(ldz[cur], lane, TN, ks * 32,
                           wn * 64 + i * 16);
#pragma unroll
      for (int j = 0; j < 2; ++j)
        b[ks][j] = tr_frag(lx[cur], lane, TK, ks * 32,
                           wk * 32 + j * 16);
    }
    lds_wait_all();
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
      for (int i = 0; i < 4; ++i) lds_pin(a[ks][i]);
#pragma unroll
      for (int j = 0; j < 2; ++j) lds_pin(b[ks][j]);
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[ks][i], b[ks][j], acc[i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    // counted wait: next tile's 3 staging calls must have landed; the
    // tile-after-next's 3 calls (just issued) may stay in flight
    if (mt + 2 < mt1)
      asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  const int c_sub_row = (lane / 16) * 4;
  const int c_col = lane % 16;
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int col = k0 + wk * 32 + j * 16 + c_col;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int row = n0 + wn * 64 + i * 16 + c_sub_row + rr;
        if (splits > 1)
          atomicAdd(&dW[(long long)row * K + col], acc[i][j][rr]);
        else
          dW[(long long)row * K + col] = acc[i][j][rr];
      }
    }
}

// ------------------------------------------------------------ harness

static short host_f2bf(float f) {
  union { float f; unsigned u; } c;
  c.f = f;
  unsigned lsb = (c.u >> 16) & 1u;
  c.u += 0x7fffu + lsb;
  return (short)(c.u >> 16);
}
static float host_b2f(short s) {
  union { float f; unsigned u; } c;
  c.u = ((unsigned)(unsigned short)s) << 16;
  return c.f;
}

static int run_case(long long M, int N, int K, bool check, int iters) {
  std::vector<short> hZ((size_t)M * N), hX((size_t)M * K);
  srand(17);
  for (auto& v : hZ) v = host_f2bf((rand() % 2000 - 1000) / 1000.0f);
  for (auto& v : hX) v = host_f2bf((rand() % 2000 - 1000) / 1000.0f);
  short *dZ, *dX;
  float* dW;
  HIP_CHECK(hipMalloc(&dZ, hZ.size() * 2));
  HIP_CHECK(hipMalloc(&dX, hX.size() * 2));
  HIP_CHECK(hipMalloc(&dW, (size_t)N * K * 4));
  HIP_CHECK(hipMemcpy(dZ, hZ.data(), hZ.size() * 2, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(dX, hX.data(), hX.size() * 2, hipMemcpyHostToDevice));

  const int ntiles = (N / TN) * (K / TK);
  int splits = 1;
  while (ntiles * splits < 1024 && splits < 64 &&
         (long long)splits * 2 * TM <= M)
    splits *= 2;
  auto launch = [&]() {
    if (splits > 1)
      HIP_CHECK(hipMemsetAsync(dW, 0, (size_t)N * K * 4));
    hipLaunchKernelGGL(wgrad_k, dim3(ntiles * splits), dim3(THREADS), 0,
                       0, dZ, dX, dW, M, N, K, splits);
  };
  launch();
  HIP_CHECK(hipDeviceSynchronize());

  int bad = 0;
  if (check) {
    std::vector<float> ref((size_t)N * K, 0.f);
    for (long long m = 0; m < M; ++m)
      for (int n = 0; n < N; ++n) {
        const float z = host_b2f(hZ[m * N + n]);
        for (int k = 0; k < K; ++k)
          ref[(size_t)n * K + k] += z * host_b2f(hX[m * K + k]);
      }
    std::vector<float> hW((size_t)N * K);
    HIP_CHECK(hipMemcpy(hW.data(), dW, hW.size() * 4,
                        hipMemcpyDeviceToHost));
    for (size_t i = 0; i < hW.size(); ++i)
      if (fabsf(hW[i] - ref[i]) > 5e-2f + 2e-2f * fabsf(ref[i])) {
        if (bad < 5)
          printf("  mismatch [%zu]: got %f want %f\n", i, hW[i], ref[i]);
        ++bad;
      }
    printf("wgrad refcheck M=%lld N=%d K=%d splits=%d: %s (%d bad)\n",
           M, N, K, splits, bad ? "FAIL" : "ok", bad);
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
    printf("wgrad perf M=%lld N=%d K=%d splits=%d: %.1f us, %.0f TF\n",
           M, N, K, splits, ms / iters * 1000,
           2.0 * M * N * K * iters / (ms / 1e3) / 1e12);
  }
  HIP_CHECK(hipFree(dZ));
  HIP_CHECK(hipFree(dX));
  HIP_CHECK(hipFree(dW));
  return bad;
}

int main() {
  int bad = 0;
  bad += run_case(256, 256, 256, true, 1);
  bad += run_case(512, 512, 256, true, 1);
  bad += run_case(4096, 256, 256, true, 1);  // forces splits > 1
  if (bad) {
    printf("WGRAD REFCHECK FAILED\n");
    return 1;
  }
  // BERT wgrad shapes (bs64 s512: M=32768); Tensile bar: 237/221/205/149 us
  run_case(32768, 3072, 768, false, 10);
  run_case(32768, 768, 3072, false, 10);
  run_case(32768, 2304, 768, false, 10);
  run_case(32768, 768, 768, false, 10);
  return 0;
}
