// Hand-written MFMA bf16 GEMM with fused bias(+GELU) epilogue
// (transformer GEMM+bias+GELU block).
//
//   C[M,N] = act(A[M,K] @ W[N,K]^T + bias[N]),  optional Z = pre-act
//
// Both operands are K-major (A row-major, W = torch Linear weight
// [N,K]), so A- and B-fragment loads share one LDS access pattern.
// The dgrad GEMM (dX = dZ @ W) reuses this kernel with the transposed
// weight (transpose_bf16 below) as the B operand.
//
// Structure — the measured round-2 winner (experimental/gemm256_v5.hip
// "G16"; probe ladder in profiles/gemm_v*.log):
//   - 256x256 output tile, K-step 64, 1024 threads = 16 waves in a
//     4x4 grid; per-wave output 64x64 = 4x4 fragments of
//     v_mfma_f32_16x16x32_bf16. 128 VGPR/wave -> 4 waves/SIMD — the
//     occupancy lever: 8-wave/2-SIMD-wave variants plateau ~1100 TF
//     while this reaches 1201 TF @4096^3 (hipBLASLt: 1450); on the
//     BERT shapes 769-979 TF, and with the fused epilogue it replaces
//     GEMM + separate bias_gelu passes.
//   - double-buffered LDS (2 x (A 256x64 + B 256x64) = 128 KiB,
//     dynamic), staged via global_load_lds width 16 with the st_16x32
//     XOR swizzle applied by pre-permuting the per-lane GLOBAL source
//     (PMC: SQ_LDS_BANK_CONFLICT 1.18M -> 0)
//   - s_setprio(1) around the MFMA cluster
//   - XCD-aware bijective blockIdx swizzle (8 XCDs, private L2s)
//   - epilogue: bias add (+ exact-erf GELU) in fp32, bf16 store,
//     optional pre-activation Z for the backward pass
//   - edges: A/B row indices clamp to the last valid row (out-of-range
//     rows compute garbage in out-of-range outputs only), C/Z stores
//     are predicated; host requires only K % 64 == 0.
//
// MFMA fragment layout for mfma_f32_16x16x32_bf16 (HW-verified):
// A-operand lane l holds A[row=l%16][k=8*(l/16)..+8]; B-operand lane l
// holds B[k=8*(l/16)..+8][col=l%16]; C/D lane l reg r holds
// C[row=(l/16)*4+r][col=l%16].

#include "common.hip.h"
#include "kernels.h"

namespace {

constexpr int kBM = 256;
constexpr int kBN = 256;
constexpr int kBK = 64;
constexpr int kThreads = 1024;
constexpr size_t kLds = 2 * 2 * (size_t)kBM * kBK * sizeof(short);

typedef float float4x __attribute__((ext_vector_type(4)));
typedef short bf16x8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float gelu_erf(float z) {
  return 0.5f * z * (1.f + erff(z * 0.70710678118654752f));
}

// st_16x32 XOR swizzle on a [row][64] bf16 tile byte offset (rows are
// 128 B): XOR byte bits 4-6 with row bits 0-2. Involution; keeps 16 B
// chunks intact; kills the 16-way ds_read_b128 bank conflict.
__device__ __forceinline__ int swz(int byte_off) {
  return byte_off ^ (((byte_off >> 7) & 7) << 4);
}

// act: 0 = identity(+bias), 1 = GELU(+bias)
template <int ACT, bool SAVE_Z>
__global__ __launch_bounds__(kThreads) void gemm_bias_act_k(
    const short* __restrict__ A, const short* __restrict__ W,
    const float* __restrict__ bias, short* __restrict__ C,
    short* __restrict__ Z, int M, int N, int K) {
  extern __shared__ short lds[];

  // XCD-aware bijective swizzle (guide §5, m204 formula).
  const int nwg = gridDim.x;
  const int q = nwg / 8, r = nwg % 8;
  const int xcd = blockIdx.x % 8, idx = blockIdx.x / 8;
  const int wgid =
      (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  const int ntn = (N + kBN - 1) / kBN;
  const long long a_row0 = (long long)(wgid / ntn) * kBM;
  const long long b_row0 = (long long)(wgid % ntn) * kBN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 2;  // 0..3 -> output rows [wr*64, +64)
  const int wc = wave & 3;   // 0..3 -> output cols [wc*64, +64)
  const int frag_row = lane % 16;
  const int frag_k = (lane / 16) * 8;

  float4x acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  auto bufA = [&](int b) { return lds + (size_t)b * 2 * kBM * kBK; };
  auto bufB = [&](int b) { return lds + ((size_t)b * 2 + 1) * kBM * kBK; };

  // Stage one 256x64 operand tile: 2 cooperative global_load_lds calls
  // x 1024 threads x 16 B. LDS write is linear (wave-uniform base +
  // lane*16); the per-lane GLOBAL source is pre-permuted with the same
  // swizzle the ds_reads apply. Rows clamp to the operand's last row.
  auto stage = [&](short* ldst, const short* g, long long row0, int k0,
                   int ld, int nrows) {
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      const int e_base = (s * kThreads + tid) * 8;
      const int e = swz(e_base * 2) / 2;
      int row = e / kBK;
      const int kk = e % kBK;
      long long grow = row0 + row;
      if (grow >= nrows) grow = nrows - 1;
      const short* gp = g + grow * (long long)ld + k0 + kk;
      short* lp = ldst + ((s * kThreads + (tid & ~63)) * 8);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gp,
          (__attribute__((address_space(3))) unsigned int*)lp, 16, 0, 0);
    }
  };
  auto ld_frag = [&](const short* ldst, int row, int kk) -> bf16x8 {
    const int byte = swz((row * kBK + kk) * 2);
    return *(const bf16x8*)((const char*)ldst + byte);
  };

  stage(bufA(0), A, a_row0, 0, K, M);
  stage(bufB(0), W, b_row0, 0, K, N);
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();

  int cur = 0;
  const int ntiles = K / kBK;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {
      stage(bufA(cur ^ 1), A, a_row0, (t + 1) * kBK, K, M);
      stage(bufB(cur ^ 1), W, b_row0, (t + 1) * kBK, K, N);
    }
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a[4], b[4];
#pragma unroll
      for (int i = 0; i < 4; ++i)
        a[i] = ld_frag(bufA(cur), wr * 64 + i * 16 + frag_row,
                       ks * 32 + frag_k);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        b[j] = ld_frag(bufB(cur), wc * 64 + j * 16 + frag_row,
                       ks * 32 + frag_k);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
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

  // Epilogue: fp32 bias+activation per fragment
  // (C[row=(lane/16)*4+r][col=lane%16]), then LDS-staged COALESCED
  // writeout — the direct per-lane stores are 2-byte column scatters
  // (measured ~2x kernel time on the save-Z FFN shape); staging the
  // 256x256 bf16 tile in the now-free K-loop LDS turns the global
  // writes into row-contiguous bf16x8 stores.
  const int c_sub_row = (lane / 16) * 4;
  const int c_col = lane % 16;
  float bv[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const long long col = b_row0 + wc * 64 + j * 16 + c_col;
    bv[j] = (bias && col < N) ? bias[col] : 0.f;
  }

  auto stage_out = [&](bool pre_act) {
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int col = wc * 64 + j * 16 + c_col;
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          const int row = wr * 64 + i * 16 + c_sub_row + rr;
          const float z = acc[i][j][rr] + bv[j];
          lds[row * kBN + col] =
              f2bf(!pre_act && ACT == 1 ? gelu_erf(z) : z);
        }
    }
    __builtin_amdgcn_s_barrier();
  };
  auto store_out = [&](short* __restrict__ dst) {
    // 1024 threads x 16 B = 32 rows per sweep; 8 sweeps for 256 rows
#pragma unroll
    for (int s = 0; s < 8; ++s) {
      const int row = s * 32 + tid / 32;
      const long long grow = a_row0 + row;
      if (grow >= M) continue;
      const int col = (tid % 32) * 8;
      const long long gcol = b_row0 + col;
      const bf16x8 v = *(const bf16x8*)&lds[row * kBN + col];
      if (gcol + 8 <= N) {
        *(bf16x8*)&dst[grow * N + gcol] = v;
      } else {
        for (int u = 0; u < 8 && gcol + u < N; ++u)
          dst[grow * N + gcol + u] = v[u];
      }
    }
  };

  stage_out(false);
  store_out(C);
  if (SAVE_Z) {
    stage_out(true);
    store_out(Z);
  }
}

// ---------------------------------------------------------------------
// bf16 matrix transpose (for the dgrad GEMM's W^T operand): 64x64
// tiles through LDS, vectorized 8-wide reads, padded LDS rows to dodge
// bank conflicts. Weight-sized inputs (a few MB) — bandwidth-trivial.
// ---------------------------------------------------------------------

constexpr int kTT = 64;  // transpose tile

__global__ __launch_bounds__(256) void transpose_bf16_k(
    const short* __restrict__ X, short* __restrict__ Y, int R, int C) {
  __shared__ short tile[kTT][kTT + 8];
  const int tr = blockIdx.x % ((R + kTT - 1) / kTT);
  const int tc = blockIdx.x / ((R + kTT - 1) / kTT);
  const int r0 = tr * kTT, c0 = tc * kTT;
  // load: 256 threads cover 32 rows x 64 cols per sweep; 2 sweeps
#pragma unroll
  for (int s = 0; s < 2; ++s) {
    const int rr = s * 32 + threadIdx.x / 8;
    const int cc = (threadIdx.x % 8) * 8;
    if (r0 + rr < R) {
#pragma unroll
      for (int u = 0; u < 8; ++u)
        tile[rr][cc + u] =
            (c0 + cc + u < C) ? X[(long long)(r0 + rr) * C + c0 + cc + u]
                              : (short)0;
    }
  }
  __syncthreads();
  // store transposed: 32 output rows (= input cols) per sweep
#pragma unroll
  for (int s = 0; s < 2; ++s) {
    const int cc = s * 32 + threadIdx.x / 8;  // output row = input col
    const int rr = (threadIdx.x % 8) * 8;
    if (c0 + cc < C) {
#pragma unroll
      for (int u = 0; u < 8; ++u)
        if (r0 + rr + u < R)
          Y[(long long)(c0 + cc) * R + r0 + rr + u] = tile[rr + u][cc];
    }
  }
}

}  // namespace

void launch_gemm_bias_act(const short* A, const short* W,
                          const float* bias, short* C, short* Z, int M,
                          int N, int K, int act, hipStream_t stream) {
  const int grid =
      ((M + kBM - 1) / kBM) * ((N + kBN - 1) / kBN);
  static bool lds_set = false;
  if (!lds_set) {
    (void)hipFuncSetAttribute((const void*)&gemm_bias_act_k<0, false>,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              (int)kLds);
    (void)hipFuncSetAttribute((const void*)&gemm_bias_act_k<1, false>,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              (int)kLds);
    (void)hipFuncSetAttribute((const void*)&gemm_bias_act_k<1, true>,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              (int)kLds);
    lds_set = true;
  }
  if (act == 1) {
    if (Z)
      hipLaunchKernelGGL((gemm_bias_act_k<1, true>), dim3(grid),
                         dim3(kThreads), kLds, stream, A, W, bias, C, Z,
                         M, N, K);
    else
      hipLaunchKernelGGL((gemm_bias_act_k<1, false>), dim3(grid),
                         dim3(kThreads), kLds, stream, A, W, bias, C, Z,
                         M, N, K);
  } else {
    hipLaunchKernelGGL((gemm_bias_act_k<0, false>), dim3(grid),
                       dim3(kThreads), kLds, stream, A, W, bias, C, Z, M,
                       N, K);
  }
}

void launch_transpose_bf16(const short* X, short* Y, int R, int C,
                           hipStream_t stream) {
  const int grid = ((R + kTT - 1) / kTT) * ((C + kTT - 1) / kTT);
  hipLaunchKernelGGL(transpose_bf16_k, dim3(grid), dim3(256), 0, stream,
                     X, Y, R, C);
}

bool gemm_bias_act_supported(int M, int N, int K) {
  // edges are clamped/predicated in-kernel; only the K-loop step is a
  // hard requirement
  return K % kBK == 0 && M >= 1 && N >= 1;
}
