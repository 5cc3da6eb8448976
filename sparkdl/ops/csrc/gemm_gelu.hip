// Hand-written MFMA bf16 GEMM with fused bias(+GELU) epilogue
// (transformer GEMM+bias+GELU block).
//
//   C[M,N] = act(A[M,K] @ W[N,K]^T + bias[N])
//
// Both operands are K-major (A row-major, W = torch Linear weight
// [N,K]), so A- and B-fragment loads share one LDS access pattern.
//
// Structure (the verified m97-class schedule from the CDNA4 guide §5):
//   - 128x128 output tile, K-step 32, 256 threads = 4 waves in 2x2,
//     each wave owns a 64x64 quadrant = 4x4 fragments of
//     v_mfma_f32_16x16x32_bf16 accumulating in AGPRs
//   - global->LDS staging via __builtin_amdgcn_global_load_lds width 16
//     (wave-uniform LDS base + lane*16; LDS stays linear row-major
//     [128][32] — per-lane global addresses are computed to match)
//   - 8 x ds_read_b128 fragment loads + 16 MFMA per K-step
//   - XCD-aware bijective blockIdx swizzle (8 XCDs, private L2s)
//   - epilogue: bias add (+ exact-erf GELU) in fp32, bf16 store; the
//     pre-activation z is optionally saved for the backward pass.
//
// MFMA fragment layout for mfma_f32_16x16x32_bf16 (HW-verified in the
// guide): A-operand lane l holds A[row=l%16][k=8*(l/16)..+8]; B-operand
// lane l holds B[k=8*(l/16)..+8][col=l%16]; C/D lane l reg r holds
// C[row=(l/16)*4+r][col=l%16].

#include "common.hip.h"
#include "kernels.h"

namespace {

constexpr int kBM = 128;
constexpr int kBN = 128;
constexpr int kBK = 32;
constexpr int kThreads = 256;

typedef float float4x __attribute__((ext_vector_type(4)));
typedef short bf16x8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float gelu_erf(float z) {
  return 0.5f * z * (1.f + erff(z * 0.70710678118654752f));
}

// act: 0 = identity(+bias), 1 = GELU(+bias)
template <int ACT, bool SAVE_Z>
__global__ __launch_bounds__(kThreads) void gemm_bias_act_k(
    const short* __restrict__ A, const short* __restrict__ W,
    const float* __restrict__ bias, short* __restrict__ C,
    short* __restrict__ Z, int M, int N, int K) {
  __shared__ short lA[kBM * kBK];
  __shared__ short lB[kBN * kBK];

  // XCD-aware bijective swizzle (guide §5, m204 formula).
  const int nwg = gridDim.x;
  const int q = nwg / 8, r = nwg % 8;
  const int xcd = blockIdx.x % 8, idx = blockIdx.x / 8;
  const int wgid =
      (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  const int ntn = N / kBN;
  const int bm = wgid / ntn;
  const int bn = wgid % ntn;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;  // wave row (0..1)
  const int wc = wave & 1;   // wave col (0..1)

  // Staging geometry: tile = 128*32 bf16 = 4096 elems; thread loads
  // 2 x 8 elems (16 B) per tile; element base for (wave, s, lane):
  //   e = (s*256 + wave*64 + lane) * 8 ; lds byte = e*2.
  const long long a_row0 = (long long)bm * kBM;
  const long long b_row0 = (long long)bn * kBN;

  float4x acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int frag_row = lane % 16;
  const int frag_k = (lane / 16) * 8;

  for (int k0 = 0; k0 < K; k0 += kBK) {
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      const int e = (s * 256 + tid) * 8;
      const int row = e / kBK, kk = e % kBK;
      const short* ga = A + (a_row0 + row) * K + k0 + kk;
      const short* gb = W + (b_row0 + row) * K + k0 + kk;
      // wave-uniform LDS base: lane*16 bytes added by hardware
      short* la = lA + (s * 256 + wave * 64) * 8;
      short* lb = lB + (s * 256 + wave * 64) * 8;
      __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) unsigned int*)ga,
                                       (__attribute__((address_space(3))) unsigned int*)la, 16, 0, 0);
      __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) unsigned int*)gb,
                                       (__attribute__((address_space(3))) unsigned int*)lb, 16, 0, 0);
    }
    __builtin_amdgcn_s_waitcnt(0);  // vmcnt(0): staging complete
    __syncthreads();

    bf16x8 a[4], b[4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
      a[i] = *(const bf16x8*)(lA + (wr * 64 + i * 16 + frag_row) * kBK +
                              frag_k);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      b[j] = *(const bf16x8*)(lB + (wc * 64 + j * 16 + frag_row) * kBK +
                              frag_k);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[i], b[j], acc[i][j], 0, 0, 0);
    __syncthreads();
  }

  // Epilogue: C[row=(lane/16)*4+r][col=lane%16] per fragment.
  const int c_sub_row = (lane / 16) * 4;
  const int c_col = lane % 16;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const long long col = b_row0 + wc * 64 + j * 16 + c_col;
    const float bv = bias ? bias[col] : 0.f;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const long long row = a_row0 + wr * 64 + i * 16 + c_sub_row + rr;
        const float z = acc[i][j][rr] + bv;
        const float y = ACT == 1 ? gelu_erf(z) : z;
        C[row * N + col] = f2bf(y);
        if (SAVE_Z) Z[row * N + col] = f2bf(z);
      }
    }
  }
}

}  // namespace

void launch_gemm_bias_act(const short* A, const short* W,
                          const float* bias, short* C, short* Z, int M,
                          int N, int K, int act, hipStream_t stream) {
  const int grid = (M / kBM) * (N / kBN);
  if (act == 1) {
    if (Z)
      hipLaunchKernelGGL((gemm_bias_act_k<1, true>), dim3(grid),
                         dim3(kThreads), 0, stream, A, W, bias, C, Z, M,
                         N, K);
    else
      hipLaunchKernelGGL((gemm_bias_act_k<1, false>), dim3(grid),
                         dim3(kThreads), 0, stream, A, W, bias, C, Z, M,
                         N, K);
  } else {
    hipLaunchKernelGGL((gemm_bias_act_k<0, false>), dim3(grid),
                       dim3(kThreads), 0, stream, A, W, bias, C, Z, M, N,
                       K);
  }
}

bool gemm_bias_act_supported(int M, int N, int K) {
  return M % kBM == 0 && N % kBN == 0 && K % kBK == 0;
}
