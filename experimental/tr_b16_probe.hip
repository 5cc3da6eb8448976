// EXPERIMENTAL — ds_read_b64_tr_b16 semantics probe (gfx950).
//
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 \
//       experimental/tr_b16_probe.hip -o /tmp/trp && /tmp/trp
//
// Purpose: pin down the exact lane/element mapping of the hardware
// transpose read so the round-3 wgrad kernel can replace its scalar
// transpose-scatter staging (the measured bound: 336 TF vs Tensile's
// ~650) with linear global_load_lds staging + transpose fragment
// reads.
//
// Method: fill LDS with self-identifying values (encode (row, col) of
// a [32][64] bf16 tile as small integers exactly representable in
// bf16), issue the transpose read with a few candidate per-lane
// address schemes, and print the (lane, elem) -> (row, col) map.
//
// MEASURED RESULT (MI355X, scheme 0: addr = lane*8):
//   lane l elems j=0..3 = shorts {l, l+16, l+32, l+48} of the 128-byte
//   region the 16-lane group's addresses cover. I.e. per 16-lane
//   group, viewing the 64 gathered shorts as a row-major [4][16]
//   matrix M (lane l's own b64 supplies M[l/4][4*(l%4)..+4]), the
//   instruction delivers COLUMN (l&15) to lane l with elems j = the 4
//   rows — a hardware [4][16] -> [16][4] bf16 transpose where each
//   lane's address independently selects its source b64, so the
//   source subtile may be strided (e.g. a [4 m][16 n] patch of a
//   row-major [64][256] tile at addr = ((m0+l/4)*256 + n0+4*(l%4))*2).
//   wgrad A-fragment recipe (round 3): group g reads subtiles at
//   m0 = 8g and 8g+4 -> lane l holds n = n0+(l&15), m = 8g..8g+8,
//   exactly the MFMA A-operand layout, from LINEARLY staged LDS.
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

typedef short b16x4 __attribute__((ext_vector_type(4)));

#define HIP_CHECK(x)                                                     \
  do {                                                                   \
    hipError_t e = (x);                                                  \
    if (e != hipSuccess) {                                               \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(e), __FILE__,  \
             __LINE__);                                                  \
      exit(1);                                                           \
    }                                                                    \
  } while (0)

__device__ __forceinline__ short enc(int row, int col) {
  // value = row*100 + col, exactly representable in bf16 for < 6500
  union { float f; unsigned u; } c;
  c.f = (float)(row * 100 + col);
  return (short)(c.u >> 16);
}
__device__ __forceinline__ int dec(short s) {
  union { float f; unsigned u; } c;
  c.u = ((unsigned)(unsigned short)s) << 16;
  return (int)c.f;
}

__global__ void tr_probe_k(int* out, int scheme) {
  __shared__ short lds[32 * 64];
  const int lane = threadIdx.x & 63;
  // fill: [32 rows][64 cols] row-major
  for (int e = threadIdx.x; e < 32 * 64; e += blockDim.x)
    lds[e] = enc(e / 64, e % 64);
  __syncthreads();

  // candidate per-lane base addresses (bytes)
  int addr;
  const int lg = lane & 15;
  switch (scheme) {
    case 0: addr = lane * 8; break;                  // linear b64
    case 1:  // wgrad A-frag scheme, col0=0: [4][16] subtile rows
             // m0+(lg>>2), cols 4*(lg&3), stride 64; m0 = 8*(lane>>4)
      addr = (((lane >> 4) * 8 + (lg >> 2)) * 64 + 4 * (lg & 3)) * 2;
      break;
    case 2:  // same with col0 = 16
      addr = (((lane >> 4) * 8 + (lg >> 2)) * 64 + 16 + 4 * (lg & 3)) * 2;
      break;
    default:  // same with col0 = 32
      addr = (((lane >> 4) * 8 + (lg >> 2)) * 64 + 32 + 4 * (lg & 3)) * 2;
      break;
  }
  b16x4 v;
  // address = LDS byte offset of the array base + per-lane offset
  const unsigned base = (unsigned)(unsigned long long)(
      (__attribute__((address_space(3))) short*)lds);
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(v)
               : "v"(base + (unsigned)addr)
               : "memory");
  if (threadIdx.x < 64) {
#pragma unroll
    for (int j = 0; j < 4; ++j) out[lane * 4 + j] = dec(v[j]);
  }
}

int main() {
  int* d;
  HIP_CHECK(hipMalloc(&d, 64 * 4 * sizeof(int)));
  int h[256];
  for (int scheme = 0; scheme < 4; ++scheme) {
    hipLaunchKernelGGL(tr_probe_k, dim3(1), dim3(64), 0, 0, d, scheme);
    hipError_t e = hipDeviceSynchronize();
    if (e != hipSuccess) {
      printf("scheme %d launch failed: %s\n", scheme,
             hipGetErrorString(e));
      continue;
    }
    HIP_CHECK(hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost));
    printf("scheme %d: value = row*100+col of a [32][64] bf16 tile\n",
           scheme);
    for (int l = 0; l < 64; l += 1) {
      printf("  lane %2d:", l);
      for (int j = 0; j < 4; ++j) printf(" %5d", h[l * 4 + j]);
      printf("\n");
    }
  }
  HIP_CHECK(hipFree(d));
  return 0;
}
