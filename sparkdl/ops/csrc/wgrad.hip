// Split-M wgrad GEMM: dW[N,K] = dZ^T[N,M] @ X[M,K] (both operands
// M-major). Tiles are staged LINEARLY by global_load_lds into a
// conflict-free blocked LDS layout ([m/4][c/16][4][16], produced by
// pre-permuting the per-lane global source) and the MFMA fragments are
// read with ds_read_b64_tr_b16 hardware transpose reads (lane
// semantics pinned by experimental/tr_b16_probe.hip). fp32 partial
// tiles combine with atomicAdd across the split-M grid.
//
// Measured 506-534 TF on the BERT wgrad shapes vs Tensile's ~650, so
// the op ships OFF by default (SPARKDL_FUSED_WGRAD=1 opts in); probe
// ladder in experimental/wgrad_v*.hip and profiles/wgrad_*.log.

#include "common.hip.h"
#include "kernels.h"

namespace {

constexpr int TN = 256;
constexpr int TK = 128;
constexpr int TM = 64;
constexpr int WTHREADS = 1024;

typedef short bf16x8w __attribute__((ext_vector_type(8)));
typedef float f32x4w __attribute__((ext_vector_type(4)));
#define bf16x8 bf16x8w
#define f32x4 f32x4w

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

__global__ __launch_bounds__(WTHREADS) void wgrad_k(
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
      const int p = (s * WTHREADS + tid) * 8;
      const int blkid = p >> 6, within = p & 63;
      const int m = (blkid / (TN >> 4)) * 4 + (within >> 4);
      const int c = (blkid % (TN >> 4)) * 16 + (within & 15);
      const short* gp = dZ + (m0 + m) * (long long)N + n0 + c;
      short* lp = ldz[buf] + ((s * WTHREADS + (tid & ~63)) * 8);
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


}  // namespace

bool wgrad_splitk_supported(long long M, int N, int K) {
  return N % TN == 0 && K % TK == 0 && M % TM == 0 && M >= TM;
}

void launch_wgrad_splitk(const short* dZ, const short* X, float* dW,
                         long long M, int N, int K, hipStream_t stream) {
  const int ntiles = (N / TN) * (K / TK);
  int splits = 1;
  while (ntiles * splits < 1024 && splits < 64 &&
         (long long)splits * 2 * TM <= M)
    splits *= 2;
  if (splits > 1)
    (void)hipMemsetAsync(dW, 0, (size_t)N * K * sizeof(float), stream);
  hipLaunchKernelGGL(wgrad_k, dim3(ntiles * splits), dim3(WTHREADS), 0,
                     stream, dZ, X, dW, M, N, K, splits);
}
