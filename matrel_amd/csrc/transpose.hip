// Materialized transpose (tiled through LDS, HBM-bandwidth-bound).
//
// Reference equivalent: MatRel's JVM block transpose (SURVEY.md §2.2 row
// "fused HIP elementwise+transpose"). Used by the executor's Transpose op
// and by SUMMA to pre-transpose B panels for the B^T GEMM kernel.
//
// 64x64 tiles; loads are 8-wide vectors along the input row (coalesced),
// stores 8-wide along the output row; the in-tile transpose happens in
// LDS with a padded stride (72 elems = 9 x 16 B -> conflict-reduced).
// Ragged edges handled with bounds checks (edge blocks of a BlockMatrix).
#include "common.h"

#define TT 64
#define TPAD 72  // lds row stride in elements

template <typename T, typename V8>
__global__ void transpose_k(T* __restrict__ out, const T* __restrict__ in,
                            long rows, long cols) {
  __shared__ T tile[TT * TPAD];
  // block covers input tile [r0, r0+64) x [c0, c0+64)
  long tr = blockIdx.x / ((cols + TT - 1) / TT);
  long tc = blockIdx.x % ((cols + TT - 1) / TT);
  long r0 = tr * TT, c0 = tc * TT;

  const u32 tx = threadIdx.x & 7;   // 8 x 8-elem slots per row
  const u32 ty = threadIdx.x >> 3;  // 32 row-pairs

  // load: rows ty, ty+32
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    long r = r0 + ty + h * 32;
    long c = c0 + tx * 8;
    if (r < rows) {
      if (c + 8 <= cols) {
        // vector LDS store: the dst (row*TPAD + tx*8) elements are
        // contiguous and 16-B aligned (TPAD*sizeof(T) is a 16-multiple)
        V8 v = *(const V8*)(in + r * cols + c);
        *(V8*)(tile + (ty + h * 32) * TPAD + tx * 8) = v;
      } else {
        for (int j = 0; j < 8 && c + j < cols; ++j)
          tile[(ty + h * 32) * TPAD + tx * 8 + j] = in[r * cols + c + j];
      }
    }
  }
  __syncthreads();

  // store: output tile [c0, c0+64) x [r0, r0+64): out[c][r] = tile[r][c]
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    long oc = c0 + ty + h * 32;  // output row = input col
    long orow = r0 + tx * 8;     // output col = input row
    if (oc < cols) {
      if (orow + 8 <= rows) {
        T v[8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          v[j] = tile[(tx * 8 + j) * TPAD + ty + h * 32];
        *(V8*)(out + oc * rows + orow) = *(const V8*)v;
      } else {
        for (int j = 0; j < 8 && orow + j < rows; ++j)
          out[oc * rows + orow + j] = tile[(tx * 8 + j) * TPAD + ty + h * 32];
      }
    }
  }
}

typedef bf16 bf16v8 __attribute__((ext_vector_type(8)));
typedef float f32v8 __attribute__((ext_vector_type(8)));

extern "C" void launch_transpose_bf16(void* out, const void* in, long rows,
                                      long cols, hipStream_t s) {
  long nt = ((rows + TT - 1) / TT) * ((cols + TT - 1) / TT);
  hipLaunchKernelGGL((transpose_k<bf16, bf16v8>), dim3((unsigned)nt),
                     dim3(256), 0, s, (bf16*)out, (const bf16*)in, rows,
                     cols);
}

extern "C" void launch_transpose_f32(float* out, const float* in, long rows,
                                     long cols, hipStream_t s) {
  hipLaunchKernelGGL(
      (transpose_k<float, f32v8>),
      dim3((unsigned)(((rows + TT - 1) / TT) * ((cols + TT - 1) / TT))),
      dim3(256), 0, s, out, in, rows, cols);
}
