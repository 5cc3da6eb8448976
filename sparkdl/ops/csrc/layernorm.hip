// Fused LayerNorm forward/backward, bf16 I/O with fp32 statistics
// (SURVEY.md §2.2 N4).
//
// Design for CDNA4: one 64-lane wavefront owns one row (BERT-base rows
// are 768/1024 wide — a wave covers a row in 1-2 vector iterations).
// Rows are reduced with in-register 64-wide butterfly shuffles — no LDS
// round-trip for the statistics. bf16 activations move as short8
// (16 B/lane, Guideline 13 — scalar bf16 loads cost 2-2.5x); all
// accumulation is fp32 (bf16 numerics parity, SURVEY.md §7 hard part 4).
//
// Backward: per-block dgamma/dbeta partials accumulate in LDS (fp32
// atomics — distinct columns per lane, cross-wave collisions only),
// written to a [part_rows, cols] workspace; a 2D-grid reduction kernel
// (row-split x column-block, coalesced) folds the workspace with global
// atomics.  part_rows is capped so each block amortizes its LDS
// accumulation over many rows.

#include "common.hip.h"
#include "kernels.h"

namespace {

constexpr int kBlock = 256;           // 4 waves = 4 rows per block-iter
constexpr int kWavesPerBlock = kBlock / WAVE;
constexpr int kVec = 8;               // bf16 elems per lane load

__global__ __launch_bounds__(kBlock) void layernorm_fwd_k(
    const short* __restrict__ x, const float* __restrict__ gamma,
    const float* __restrict__ beta, short* __restrict__ y,
    float* __restrict__ save_mean, float* __restrict__ save_rstd,
    int rows, int cols, float eps) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;

  for (int row = blockIdx.x * kWavesPerBlock + wid; row < rows;
       row += gridDim.x * kWavesPerBlock) {
    const short* xr = x + (long long)row * cols;
    short* yr = y + (long long)row * cols;

    // Pass 1: sum and sum of squares (fp32).
    float s = 0.f, ss = 0.f;
    for (int c = lane * kVec; c < cols; c += WAVE * kVec) {
      if (c + kVec <= cols) {
        const short8 v = *(const short8*)(xr + c);
#pragma unroll
        for (int j = 0; j < kVec; ++j) {
          const float f = bf2f(v[j]);
          s += f;
          ss += f * f;
        }
      } else {
        for (int cc = c; cc < cols; ++cc) {
          const float f = bf2f(xr[cc]);
          s += f;
          ss += f * f;
        }
      }
    }
    s = wave_sum(s);
    ss = wave_sum(ss);
    const float mean = s / cols;
    const float var = fmaxf(ss / cols - mean * mean, 0.f);
    const float rstd = rsqrtf(var + eps);
    if (lane == 0) {
      save_mean[row] = mean;
      save_rstd[row] = rstd;
    }

    // Pass 2: normalize + affine (x row is L1/L2-hot from pass 1).
    for (int c = lane * kVec; c < cols; c += WAVE * kVec) {
      if (c + kVec <= cols) {
        const short8 v = *(const short8*)(xr + c);
        const float4v g0 = *(const float4v*)(gamma + c);
        const float4v g1 = *(const float4v*)(gamma + c + 4);
        const float4v b0 = *(const float4v*)(beta + c);
        const float4v b1 = *(const float4v*)(beta + c + 4);
        short8 o;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          o[j] = f2bf((bf2f(v[j]) - mean) * rstd * g0[j] + b0[j]);
          o[j + 4] = f2bf((bf2f(v[j + 4]) - mean) * rstd * g1[j] + b1[j]);
        }
        *(short8*)(yr + c) = o;
      } else {
        for (int cc = c; cc < cols; ++cc)
          yr[cc] =
              f2bf((bf2f(xr[cc]) - mean) * rstd * gamma[cc] + beta[cc]);
      }
    }
  }
}

// dx = rstd * (dyg - mean(dyg) - xhat * mean(dyg * xhat)),  dyg = dy*gamma
//
// dgamma/dbeta partials accumulate in REGISTERS per thread (static
// indices — runtime-indexed ext_vector arrays spill to scratch): lane
// covers columns {lane*8 + k*512 | k < KMAX}, so a [KMAX][8] register
// tile holds the thread's column footprint.  Each WAVE writes one
// partial row of the workspace at the end; no LDS and no atomics on the
// hot path.  KMAX is a template parameter dispatched on cols (<=2048);
// wider rows fall back to the LDS-atomic variant.
template <int KMAX>
__global__ __launch_bounds__(kBlock) void layernorm_bwd_reg_k(
    const short* __restrict__ x, const short* __restrict__ dy,
    const float* __restrict__ gamma, const float* __restrict__ save_mean,
    const float* __restrict__ save_rstd, short* __restrict__ dx,
    float* __restrict__ dgamma_part, float* __restrict__ dbeta_part,
    int rows, int cols) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;

  float dg[KMAX][kVec], db[KMAX][kVec];
#pragma unroll
  for (int k = 0; k < KMAX; ++k)
#pragma unroll
    for (int j = 0; j < kVec; ++j) dg[k][j] = db[k][j] = 0.f;

  for (int row = blockIdx.x * kWavesPerBlock + wid; row < rows;
       row += gridDim.x * kWavesPerBlock) {
    const short* xr = x + (long long)row * cols;
    const short* dyr = dy + (long long)row * cols;
    short* dxr = dx + (long long)row * cols;
    const float mean = save_mean[row];
    const float rstd = save_rstd[row];

    float s1 = 0.f, s2 = 0.f;  // sum(dyg), sum(dyg * xhat)
#pragma unroll
    for (int k = 0; k < KMAX; ++k) {
      const int c = lane * kVec + k * WAVE * kVec;
      if (c + kVec <= cols) {
        const short8 xv = *(const short8*)(xr + c);
        const short8 dv = *(const short8*)(dyr + c);
        const float4v g0 = *(const float4v*)(gamma + c);
        const float4v g1 = *(const float4v*)(gamma + c + 4);
#pragma unroll
        for (int j = 0; j < kVec; ++j) {
          const float xh = (bf2f(xv[j]) - mean) * rstd;
          const float dyf = bf2f(dv[j]);
          const float dyg = dyf * (j < 4 ? g0[j] : g1[j - 4]);
          s1 += dyg;
          s2 += dyg * xh;
          dg[k][j] += dyf * xh;
          db[k][j] += dyf;
        }
      } else if (c < cols) {
        // ragged tail (cols % 8 != 0): static j, runtime bound
#pragma unroll
        for (int j = 0; j < kVec; ++j) {
          if (c + j < cols) {
            const float xh = (bf2f(xr[c + j]) - mean) * rstd;
            const float dyf = bf2f(dyr[c + j]);
            const float dyg = dyf * gamma[c + j];
            s1 += dyg;
            s2 += dyg * xh;
            dg[k][j] += dyf * xh;
            db[k][j] += dyf;
          }
        }
      }
    }
    s1 = wave_sum(s1) / cols;
    s2 = wave_sum(s2) / cols;
#pragma unroll
    for (int k = 0; k < KMAX; ++k) {
      const int c = lane * kVec + k * WAVE * kVec;
      if (c + kVec <= cols) {
        const short8 xv = *(const short8*)(xr + c);
        const short8 dv = *(const short8*)(dyr + c);
        const float4v g0 = *(const float4v*)(gamma + c);
        const float4v g1 = *(const float4v*)(gamma + c + 4);
        short8 o;
#pragma unroll
        for (int j = 0; j < kVec; ++j) {
          const float xh = (bf2f(xv[j]) - mean) * rstd;
          const float dyg = bf2f(dv[j]) * (j < 4 ? g0[j] : g1[j - 4]);
          o[j] = f2bf(rstd * (dyg - s1 - xh * s2));
        }
        *(short8*)(dxr + c) = o;
      } else if (c < cols) {
        for (int cc = c; cc < cols; ++cc) {
          const float xh = (bf2f(xr[cc]) - mean) * rstd;
          const float dyg = bf2f(dyr[cc]) * gamma[cc];
          dxr[cc] = f2bf(rstd * (dyg - s1 - xh * s2));
        }
      }
    }
  }

  // Each wave owns one workspace row: coalesced float4 stores.
  const long long wrow = (long long)blockIdx.x * kWavesPerBlock + wid;
  float* dgp = dgamma_part + wrow * cols;
  float* dbp = dbeta_part + wrow * cols;
#pragma unroll
  for (int k = 0; k < KMAX; ++k) {
    const int c = lane * kVec + k * WAVE * kVec;
    if (c + kVec <= cols) {
      *(float4v*)(dgp + c) = *(float4v*)&dg[k][0];
      *(float4v*)(dgp + c + 4) = *(float4v*)&dg[k][4];
      *(float4v*)(dbp + c) = *(float4v*)&db[k][0];
      *(float4v*)(dbp + c + 4) = *(float4v*)&db[k][4];
    } else if (c < cols) {
#pragma unroll
      for (int j = 0; j < kVec; ++j)
        if (c + j < cols) {
          dgp[c + j] = dg[k][j];
          dbp[c + j] = db[k][j];
        }
    }
  }
}

// LDS-atomic fallback for cols > 2048 (up to 8192).
extern __shared__ float ln_lds[];

__global__ __launch_bounds__(kBlock) void layernorm_bwd_k(
    const short* __restrict__ x, const short* __restrict__ dy,
    const float* __restrict__ gamma, const float* __restrict__ save_mean,
    const float* __restrict__ save_rstd, short* __restrict__ dx,
    float* __restrict__ dgamma_part, float* __restrict__ dbeta_part,
    int rows, int cols) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  float* dg_l = ln_lds;          // [cols]
  float* db_l = ln_lds + cols;   // [cols]

  for (int c = threadIdx.x; c < 2 * cols; c += kBlock) ln_lds[c] = 0.f;
  __syncthreads();

  for (int row = blockIdx.x * kWavesPerBlock + wid; row < rows;
       row += gridDim.x * kWavesPerBlock) {
    const short* xr = x + (long long)row * cols;
    const short* dyr = dy + (long long)row * cols;
    short* dxr = dx + (long long)row * cols;
    const float mean = save_mean[row];
    const float rstd = save_rstd[row];

    float s1 = 0.f, s2 = 0.f;
    for (int c = lane * kVec; c < cols; c += WAVE * kVec) {
      if (c + kVec <= cols) {
        const short8 xv = *(const short8*)(xr + c);
        const short8 dv = *(const short8*)(dyr + c);
        const float4v g0 = *(const float4v*)(gamma + c);
        const float4v g1 = *(const float4v*)(gamma + c + 4);
#pragma unroll
        for (int j = 0; j < kVec; ++j) {
          const float xh = (bf2f(xv[j]) - mean) * rstd;
          const float dyf = bf2f(dv[j]);
          const float dyg = dyf * (j < 4 ? g0[j] : g1[j - 4]);
          s1 += dyg;
          s2 += dyg * xh;
          atomicAdd(&dg_l[c + j], dyf * xh);
          atomicAdd(&db_l[c + j], dyf);
        }
      } else {
        for (int cc = c; cc < cols; ++cc) {
          const float xh = (bf2f(xr[cc]) - mean) * rstd;
          const float dyf = bf2f(dyr[cc]);
          const float dyg = dyf * gamma[cc];
          s1 += dyg;
          s2 += dyg * xh;
          atomicAdd(&dg_l[cc], dyf * xh);
          atomicAdd(&db_l[cc], dyf);
        }
      }
    }
    s1 = wave_sum(s1) / cols;
    s2 = wave_sum(s2) / cols;
    for (int c = lane * kVec; c < cols; c += WAVE * kVec) {
      if (c + kVec <= cols) {
        const short8 xv = *(const short8*)(xr + c);
        const short8 dv = *(const short8*)(dyr + c);
        const float4v g0 = *(const float4v*)(gamma + c);
        const float4v g1 = *(const float4v*)(gamma + c + 4);
        short8 o;
#pragma unroll
        for (int j = 0; j < kVec; ++j) {
          const float xh = (bf2f(xv[j]) - mean) * rstd;
          const float dyg = bf2f(dv[j]) * (j < 4 ? g0[j] : g1[j - 4]);
          o[j] = f2bf(rstd * (dyg - s1 - xh * s2));
        }
        *(short8*)(dxr + c) = o;
      } else {
        for (int cc = c; cc < cols; ++cc) {
          const float xh = (bf2f(xr[cc]) - mean) * rstd;
          const float dyg = bf2f(dyr[cc]) * gamma[cc];
          dxr[cc] = f2bf(rstd * (dyg - s1 - xh * s2));
        }
      }
    }
  }
  __syncthreads();
  // All four waves share one partial row in this variant: rows
  // kWavesPerBlock*blockIdx.x .. +3 collapse onto the first.
  const long long wrow = (long long)blockIdx.x * kWavesPerBlock;
  float* dgp = dgamma_part + wrow * cols;
  float* dbp = dbeta_part + wrow * cols;
  for (int c = threadIdx.x; c < cols; c += kBlock) {
    dgp[c] = dg_l[c];
    dbp[c] = db_l[c];
  }
  for (int r = 1; r < kWavesPerBlock; ++r)
    for (int c = threadIdx.x; c < cols; c += kBlock) {
      dgp[r * (long long)cols + c] = 0.f;
      dbp[r * (long long)cols + c] = 0.f;
    }
}

// 2D grid: x = column blocks (coalesced across threads), y = row splits.
// dgamma/dbeta must be pre-zeroed (global atomics fold the splits).
__global__ __launch_bounds__(kBlock) void ln_reduce_parts_k(
    const float* __restrict__ dgamma_part,
    const float* __restrict__ dbeta_part, float* __restrict__ dgamma,
    float* __restrict__ dbeta, int part_rows, int cols) {
  const int c = blockIdx.x * kBlock + threadIdx.x;
  if (c >= cols) return;
  float dg = 0.f, db = 0.f;
  for (int r = blockIdx.y; r < part_rows; r += gridDim.y) {
    dg += dgamma_part[(long long)r * cols + c];
    db += dbeta_part[(long long)r * cols + c];
  }
  atomicAdd(&dgamma[c], dg);
  atomicAdd(&dbeta[c], db);
}

}  // namespace

static int ln_grid(int rows) {
  int blocks = (rows + kWavesPerBlock - 1) / kWavesPerBlock;
  return min(blocks, 2048);  // grid-stride the rest (Guideline 11)
}

int layernorm_bwd_part_rows(int rows) {
  // Fewer blocks than forward: each block's LDS accumulation must
  // amortize over many rows, and the workspace stays small.
  int blocks = (rows + kWavesPerBlock - 1) / kWavesPerBlock;
  return min(blocks, 512);
}

void launch_layernorm_fwd(const short* x, const float* gamma,
                          const float* beta, short* y, float* save_mean,
                          float* save_rstd, int rows, int cols, float eps,
                          hipStream_t stream) {
  hipLaunchKernelGGL(layernorm_fwd_k, dim3(ln_grid(rows)), dim3(kBlock), 0,
                     stream, x, gamma, beta, y, save_mean, save_rstd, rows,
                     cols, eps);
}

void launch_layernorm_bwd(const short* x, const short* dy,
                          const float* gamma, const float* save_mean,
                          const float* save_rstd, short* dx,
                          float* dgamma_part, float* dbeta_part,
                          int blocks, int rows, int cols,
                          hipStream_t stream) {
  // workspace has blocks * kWavesPerBlock partial rows (one per wave)
  const int kmax = (cols + WAVE * kVec - 1) / (WAVE * kVec);
  if (kmax == 1) {
    hipLaunchKernelGGL(layernorm_bwd_reg_k<1>, dim3(blocks), dim3(kBlock),
                       0, stream, x, dy, gamma, save_mean, save_rstd, dx,
                       dgamma_part, dbeta_part, rows, cols);
  } else if (kmax == 2) {
    hipLaunchKernelGGL(layernorm_bwd_reg_k<2>, dim3(blocks), dim3(kBlock),
                       0, stream, x, dy, gamma, save_mean, save_rstd, dx,
                       dgamma_part, dbeta_part, rows, cols);
  } else if (kmax <= 4) {
    hipLaunchKernelGGL(layernorm_bwd_reg_k<4>, dim3(blocks), dim3(kBlock),
                       0, stream, x, dy, gamma, save_mean, save_rstd, dx,
                       dgamma_part, dbeta_part, rows, cols);
  } else {
    const size_t lds = 2 * (size_t)cols * sizeof(float);
    hipLaunchKernelGGL(layernorm_bwd_k, dim3(blocks), dim3(kBlock), lds,
                       stream, x, dy, gamma, save_mean, save_rstd, dx,
                       dgamma_part, dbeta_part, rows, cols);
  }
}

void launch_layernorm_reduce_parts(const float* dgamma_part,
                                   const float* dbeta_part, float* dgamma,
                                   float* dbeta, int part_rows, int cols,
                                   hipStream_t stream) {
  const dim3 grid((cols + kBlock - 1) / kBlock, min(part_rows, 32));
  hipLaunchKernelGGL(ln_reduce_parts_k, grid, dim3(kBlock), 0,
                     stream, dgamma_part, dbeta_part, dgamma, dbeta,
                     part_rows, cols);
}
