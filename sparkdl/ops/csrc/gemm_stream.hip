// Streaming tall-skinny MFMA GEMM for the ResNet 1x1 convolutions
// (SURVEY.md §2.2 N5): C[M,N] = A[M,K] @ W[N,K]^T with huge M (NHWC
// rows, up to ~1.6M at bs512) and small K/N. The 256x256 tile kernel
// is the wrong geometry here — its per-tile staging/drain overheads
// dominate when K has one or two 64-wide tiles — so this kernel:
//
//   - keeps the whole W panel for its 64-wide N-block resident in LDS
//     (K <= 256 -> at most 32 KiB), staged ONCE per workgroup,
//   - streams A fragments DIRECTLY from global memory to MFMA
//     (K-major rows are contiguous; no A staging, no A LDS),
//   - gives each workgroup a strided set of 1024-row M-blocks
//     (16 waves x 64 rows, acc 4x4/wave = 128 VGPR total budget,
//     4 waves/SIMD),
//   - writes C through a 256-row LDS bounce per chunk so the global
//     stores are row-contiguous bf16x8 (N*2 >= 128 B rows).
//
// No bias/activation: these convs are always followed by the fused
// BatchNormAct2d kernel.

#include "common.hip.h"
#include "kernels.h"

namespace {

constexpr int SBN = 64;        // N-block per workgroup
constexpr int SBM = 1024;      // M rows per iteration (16 waves x 64)
constexpr int STHREADS = 1024;
constexpr int KMAX = 256;

typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

// swizzle for a [64][K] bf16 W panel: XOR byte bits 4-6 with row bits
// 0-2 (row stride K*2 bytes, K in {64,128,256})
__device__ __forceinline__ int swz_row(int byte_off, int row_shift) {
  return byte_off ^ (((byte_off >> row_shift) & 7) << 4);
}

__global__ __launch_bounds__(STHREADS) void gemm_stream_k(
    const short* __restrict__ A, const short* __restrict__ W,
    short* __restrict__ C, long long M, int N, int K) {
  // LDS: W panel 64 x K (<= 32 KiB) + per-wave C bounce (16 x 2 KiB).
  // The bounce regions are wave-private, so the whole main loop runs
  // with NO barriers — the 16 waves free-run across m-blocks, which is
  // what hides the direct-from-global A fragment latency.
  __shared__ short lW[SBN * KMAX];
  __shared__ short lC[16 * 16 * SBN];

  const int nnb = N / SBN;
  const int nb = blockIdx.x % nnb;       // N-block
  const int widx = blockIdx.x / nnb;     // worker index within N-block
  const int nworkers = gridDim.x / nnb;
  const int row_shift = K == 64 ? 7 : (K == 128 ? 8 : 9);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int frag_row = lane % 16;
  const int frag_k = (lane / 16) * 8;
  const int c_sub_row = (lane / 16) * 4;
  const int c_col = lane % 16;

  // stage the W panel once: 64*K elements; pre-permuted source so the
  // swizzled ds_reads see linear data
  {
    const int total = SBN * K;  // 4096..16384 elements
    for (int e0 = tid * 8; e0 < total; e0 += STHREADS * 8) {
      const int e = swz_row(e0 * 2, row_shift) / 2;
      const int row = e / K, kk = e % K;
      const short* gp =
          W + ((long long)nb * SBN + row) * K + kk;
      short* lp = lW + (e0 - lane * 8);  // wave-uniform base
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp,
          (__attribute__((address_space(3))) unsigned int*)lp, 16, 0, 0);
    }
  }
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();

  const int ntk = K / 32;  // 16x16x32 K-steps
  const long long mblocks = (M + SBM - 1) / SBM;

  for (long long mb = widx; mb < mblocks; mb += nworkers) {
    const long long m0 = mb * SBM + (long long)wave * 64;
    f32x4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    for (int ks = 0; ks < ntk; ++ks) {
      bf16x8 a[4], b[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        long long row = m0 + i * 16 + frag_row;
        if (row >= M) row = M - 1;  // clamped; stores are predicated
        a[i] = *(const bf16x8*)(A + row * K + ks * 32 + frag_k);
      }
#pragma unroll
      for (int j = 0; j < 4; ++j)
        b[j] = *(const bf16x8*)((const char*)lW +
                                swz_row(((j * 16 + frag_row) * K +
                                         ks * 32 + frag_k) * 2,
                                        row_shift));
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }

    // epilogue: barrier-free per-wave bounce — 16 rows at a time
    // through this wave's private 2 KiB region, then row-contiguous
    // bf16x8 stores (same-wave LDS ops complete in order)
    short* myC = lC + wave * (16 * SBN);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int rr = 0; rr < 4; ++rr)
          myC[(c_sub_row + rr) * SBN + j * 16 + c_col] =
              f2bf(acc[i][j][rr]);
      const long long gr0 = m0 + i * 16;
#pragma unroll
      for (int s = 0; s < 2; ++s) {
        const int r = s * 8 + lane / 8;
        const int c = (lane % 8) * 8;
        const long long grow = gr0 + r;
        if (grow < M) {
          const bf16x8 v = *(const bf16x8*)&myC[r * SBN + c];
          *(bf16x8*)&C[grow * N + (long long)nb * SBN + c] = v;
        }
      }
    }
  }
}

}  // namespace

bool gemm_stream_supported(long long M, int N, int K) {
  return (K == 64 || K == 128 || K == 256) && N % SBN == 0 && M >= 1;
}

void launch_gemm_stream(const short* A, const short* W, short* C,
                        long long M, int N, int K, hipStream_t stream) {
  const int nnb = N / SBN;
  // enough workers to fill the chip; cap so small-M launches don't
  // spin empty workgroups
  long long mblocks = (M + SBM - 1) / SBM;
  int workers = (int)(mblocks < 1024 ? mblocks : 1024);
  if (workers < 1) workers = 1;
  while ((long long)workers * nnb < 1024 && workers < mblocks)
    ++workers;
  hipLaunchKernelGGL(gemm_stream_k, dim3(nnb * workers), dim3(STHREADS),
                     0, stream, A, W, C, M, N, K);
}
