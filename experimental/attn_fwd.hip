// EXPERIMENTAL — not built into sparkdl._C; standalone flash-attention
// FORWARD probe (NOTES-round2.md §2). Compile + run on an MI355X:
//
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 \
//       experimental/attn_fwd.hip -o /tmp/attn && /tmp/attn
//
// O = softmax(Q K^T / sqrt(D)) V,  bf16 in/out, fp32 accumulation,
// online (flash) softmax, no mask / no dropout, D = 64.
//
// Correctness-first structure built ONLY from fragment paths already
// HW-verified by the GEMM probe (mfma_f32_16x16x32_bf16 with both
// operands K-major and the C/D layout col=lane%16, row=(lane/16)*4+r):
//
//   - one workgroup per (batch*head, 64-row Q tile); 4 waves, each
//     owning 16 Q rows. Q fragments live in registers for the whole
//     kernel (Q pre-scaled by 1/sqrt(D)).
//   - per 64-key KV tile: stage K [key][d] linearly into LDS
//     (global_load_lds width 16) and V TRANSPOSED into LDS as
//     Vt [d][key] (scalar ds writes — a known v1 cost; the tr-read
//     instruction replaces this in the optimization ladder).
//   - S-tile = QK^T with the verified GEMM fragment loads (contraction
//     over d: A = Q[q][d], B-source = K[key][d], both K-major).
//   - online softmax in registers: each lane holds 4 q-rows x 1 column
//     per 16x16 fragment; the row statistics reduce over the 16-lane
//     column group with __shfl_xor(1,2,4,8).
//   - P routed through LDS ([q][key] bf16) so the PV product is again
//     the verified pattern (contraction over key: A = P[q][key],
//     B-source = Vt[d][key]).
//
// Known optimization ladder from here (all measured on this chip class
// by the CDNA4 guide): K-tile XOR swizzle, swapped QK^T + in-register
// P redistribution (cvt_pk + permlane32_swap), deferred-max rescale,
// async staging split, ds_read_b64_tr_b16 for V. aotriton's fwd on
// this shape measures ~93 TF; the plain-HIP ladder reaches ~900 TF.

#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <vector>

typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define HIP_CHECK(x)                                                    \
  do {                                                                  \
    hipError_t e = (x);                                                 \
    if (e != hipSuccess) {                                              \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(e), __FILE__, \
             __LINE__);                                                 \
      exit(1);                                                          \
    }                                                                   \
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

constexpr int D = 64;        // head dim
constexpr int BQ = 64;       // Q rows per workgroup (16 per wave)
constexpr int BKV = 64;      // KV tile
constexpr int THREADS = 256; // 4 waves

__global__ __launch_bounds__(THREADS) void attn_fwd_k(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, short* __restrict__ O, int S) {
  // LDS: K tile [64][64] + Vt tile [64][64] + P tile [64][64], bf16
  __shared__ short lK[BKV * D];
  __shared__ short lVt[D * BKV];
  __shared__ short lP[BQ * BKV];

  const int bh = blockIdx.x;        // batch*head index
  const int qt = blockIdx.y;        // Q tile index
  const long long base = (long long)bh * S * D;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;        // wave owns q rows [wave*16, +16)

  const int frag_row = lane % 16;
  const int frag_k = (lane / 16) * 8;
  const int c_sub_row = (lane / 16) * 4;
  const int c_col = lane % 16;

  // Q fragments in registers, pre-scaled by 1/sqrt(D). Layout matches
  // the A-operand of mfma_16x16x32 with contraction over d:
  // lane holds Q[q0 + lane%16][ks*32 + (lane/16)*8 .. +8].
  const float qscale = rsqrtf((float)D);
  bf16x8 qf[2];
  {
    const int qrow = qt * BQ + wave * 16 + frag_row;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 raw = *(const bf16x8*)(Q + base + (long long)qrow * D +
                                    ks * 32 + frag_k);
#pragma unroll
      for (int j = 0; j < 8; ++j) raw[j] = f2bf(bf2f(raw[j]) * qscale);
      qf[ks] = raw;
    }
  }

  // online-softmax state per lane: 4 q rows
  float m_run[4], l_run[4];
  f32x4 o_acc[4];  // [d-frag] x 4 rows
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -1e30f;
    l_run[r] = 0.f;
  }
#pragma unroll
  for (int df = 0; df < 4; ++df) o_acc[df] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = S / BKV;
  for (int t = 0; t < ntiles; ++t) {
    // ---- stage K tile linearly: 64x64 bf16 = 8 KiB = 256 thr x 2x16B
    {
      const short* kt = K + base + (long long)t * BKV * D;
#pragma unroll
      for (int s = 0; s < 2; ++s) {
        const int e = (s * THREADS + tid) * 8;
        const short* gp = kt + e;
        short* lp = lK + (s * THREADS + (tid & ~63)) * 8;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)gp,
            (__attribute__((address_space(3))) unsigned int*)lp, 16, 0,
            0);
      }
      // ---- stage V tile transposed: thread reads V[key][d0..+8],
      // writes Vt[d][key] (8 scalar LDS writes)
      const short* vt = V + base + (long long)t * BKV * D;
#pragma unroll
      for (int s = 0; s < 2; ++s) {
        const int e = (s * THREADS + tid) * 8;  // element in [key][d]
        const int key = e / D, d0 = e % D;
        const bf16x8 v = *(const bf16x8*)(vt + key * D + d0);
#pragma unroll
        for (int j = 0; j < 8; ++j) lVt[(d0 + j) * BKV + key] = v[j];
      }
    }
    __builtin_amdgcn_s_waitcnt(0);
    __syncthreads();

    // ---- S = Q K^T : per wave 16 q x 64 keys = 4 fragments
    f32x4 s_acc[4];
#pragma unroll
    for (int kf = 0; kf < 4; ++kf) {
      s_acc[kf] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const bf16x8 bf = *(const bf16x8*)(
            lK + (kf * 16 + frag_row) * D + ks * 32 + frag_k);
        s_acc[kf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qf[ks], bf, s_acc[kf], 0, 0, 0);
      }
    }

    // ---- online softmax: lane holds rows (lane/16)*4+r, col c_col+16f
    float pmax[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = s_acc[0][r];
#pragma unroll
      for (int kf = 1; kf < 4; ++kf) mx = fmaxf(mx, s_acc[kf][r]);
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, 64));
      pmax[r] = mx;
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float m_new = fmaxf(m_run[r], pmax[r]);
      const float scale = __expf(m_run[r] - m_new);
      float rowsum = 0.f;
#pragma unroll
      for (int kf = 0; kf < 4; ++kf) {
        const float p = __expf(s_acc[kf][r] - m_new);
        s_acc[kf][r] = p;  // reuse as P
        rowsum += p;
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        rowsum += __shfl_xor(rowsum, off, 64);
      l_run[r] = l_run[r] * scale + rowsum;
      m_run[r] = m_new;
#pragma unroll
      for (int df = 0; df < 4; ++df) o_acc[df][r] *= scale;
    }

    // ---- P to LDS [q][key] bf16 (verified A-operand source layout).
    // lP rows are wave-private (each wave writes and reads only its own
    // 16 q rows), so no barrier is needed around the P round-trip.
#pragma unroll
    for (int kf = 0; kf < 4; ++kf)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        lP[(wave * 16 + c_sub_row + r) * BKV + kf * 16 + c_col] =
            f2bf(s_acc[kf][r]);
    __syncthreads();

    // ---- O += P V : contraction over key; B-source Vt[d][key]
#pragma unroll
    for (int df = 0; df < 4; ++df) {
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const bf16x8 af = *(const bf16x8*)(
            lP + (wave * 16 + frag_row) * BKV + ks * 32 + frag_k);
        const bf16x8 bf = *(const bf16x8*)(
            lVt + (df * 16 + frag_row) * BKV + ks * 32 + frag_k);
        o_acc[df] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af, bf, o_acc[df], 0, 0, 0);
      }
    }
    __syncthreads();  // before next tile overwrites lK/lVt
  }

  // ---- epilogue: O[q][d] = o_acc / l
  const int qrow0 = qt * BQ + wave * 16;
#pragma unroll
  for (int df = 0; df < 4; ++df)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = qrow0 + c_sub_row + r;
      O[base + (long long)qrow * D + df * 16 + c_col] =
          f2bf(o_acc[df][r] / l_run[r]);
    }
}

// ---------------------------------------------------------------------
// Harness
// ---------------------------------------------------------------------

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

static void cpu_attn(const std::vector<short>& Q,
                     const std::vector<short>& K,
                     const std::vector<short>& V, std::vector<float>& O,
                     int BH, int S) {
  const float sc = 1.f / sqrtf((float)D);
  std::vector<float> row(S);
  for (int bh = 0; bh < BH; ++bh) {
    const size_t b = (size_t)bh * S * D;
    for (int q = 0; q < S; ++q) {
      float mx = -1e30f;
      for (int k = 0; k < S; ++k) {
        float s = 0;
        for (int d = 0; d < D; ++d)
          s += host_b2f(Q[b + q * D + d]) * host_b2f(K[b + k * D + d]);
        row[k] = s * sc;
        mx = fmaxf(mx, row[k]);
      }
      float l = 0;
      for (int k = 0; k < S; ++k) {
        row[k] = expf(row[k] - mx);
        l += row[k];
      }
      for (int d = 0; d < D; ++d) {
        float o = 0;
        for (int k = 0; k < S; ++k)
          o += row[k] * host_b2f(V[b + k * D + d]);
        O[b + q * D + d] = o / l;
      }
    }
  }
}

static int run_attn(int BH, int S, bool check, int iters) {
  std::vector<short> hQ((size_t)BH * S * D), hK(hQ.size()), hV(hQ.size());
  srand(7);
  for (auto& v : hQ) v = host_f2bf((rand() % 2000 - 1000) / 1000.0f);
  for (auto& v : hK) v = host_f2bf((rand() % 2000 - 1000) / 1000.0f);
  for (auto& v : hV) v = host_f2bf((rand() % 2000 - 1000) / 1000.0f);

  short *dQ, *dK, *dV, *dO;
  HIP_CHECK(hipMalloc(&dQ, hQ.size() * 2));
  HIP_CHECK(hipMalloc(&dK, hK.size() * 2));
  HIP_CHECK(hipMalloc(&dV, hV.size() * 2));
  HIP_CHECK(hipMalloc(&dO, hQ.size() * 2));
  HIP_CHECK(hipMemcpy(dQ, hQ.data(), hQ.size() * 2,
                      hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(dK, hK.data(), hK.size() * 2,
                      hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(dV, hV.data(), hV.size() * 2,
                      hipMemcpyHostToDevice));

  const dim3 grid(BH, S / BQ);
  hipLaunchKernelGGL(attn_fwd_k, grid, dim3(THREADS), 0, 0, dQ, dK, dV,
                     dO, S);
  HIP_CHECK(hipDeviceSynchronize());

  int bad = 0;
  if (check) {
    std::vector<float> ref((size_t)BH * S * D);
    cpu_attn(hQ, hK, hV, ref, BH, S);
    std::vector<short> hO(hQ.size());
    HIP_CHECK(hipMemcpy(hO.data(), dO, hO.size() * 2,
                        hipMemcpyDeviceToHost));
    for (size_t i = 0; i < hO.size(); ++i) {
      const float got = host_b2f(hO[i]), want = ref[i];
      if (fabsf(got - want) > 3e-2f + 3e-2f * fabsf(want)) {
        if (bad < 5)
          printf("  mismatch [%zu]: got %f want %f\n", i, got, want);
        ++bad;
      }
    }
    printf("attn refcheck BH=%d S=%d: %s (%d bad)\n", BH, S,
           bad ? "FAIL" : "ok", bad);
  } else {
    hipEvent_t e0, e1;
    HIP_CHECK(hipEventCreate(&e0));
    HIP_CHECK(hipEventCreate(&e1));
    HIP_CHECK(hipEventRecord(e0));
    for (int it = 0; it < iters; ++it)
      hipLaunchKernelGGL(attn_fwd_k, grid, dim3(THREADS), 0, 0, dQ, dK,
                         dV, dO, S);
    HIP_CHECK(hipEventRecord(e1));
    HIP_CHECK(hipEventSynchronize(e1));
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
    const double tf =
        4.0 * BH * (double)S * S * D * iters / (ms / 1e3) / 1e12;
    printf("attn perf BH=%d S=%d: %.3f ms/iter, %.0f TFLOP/s\n", BH, S,
           ms / iters, tf);
  }
  HIP_CHECK(hipFree(dQ));
  HIP_CHECK(hipFree(dK));
  HIP_CHECK(hipFree(dV));
  HIP_CHECK(hipFree(dO));
  return bad;
}

int main() {
  int bad = 0;
  bad += run_attn(2, 128, true, 1);
  bad += run_attn(1, 256, true, 1);
  if (bad) {
    printf("ATTN REFCHECK FAILED\n");
    return 1;
  }
  run_attn(64 * 12, 512, false, 20);   // the BERT bench shape
  run_attn(16 * 64, 2048, false, 10);
  return 0;
}
