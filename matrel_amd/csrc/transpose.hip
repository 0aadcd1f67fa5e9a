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

// ---- fused elementwise + transpose: out = (a OP b)^T in ONE pass -------
// (SURVEY.md §2.2 "fused HIP elementwise+transpose kernel(s)"). Both ops
// are HBM-bound; fusing them turns {2R+1W ewise} + {1R+1W transpose} =
// 5 streams into 2R+1W = 3 (a ~40% traffic cut for (A op B)^T plans).
// The op applies during the LDS-tile store; the transpose side is
// identical to transpose_k.

__device__ __forceinline__ float ewt_apply(int op, float a, float b) {
  switch (op) {
    case 0: return a + b;            // EW_ADD (codes match elementwise.hip)
    case 1: return a - b;            // EW_SUB
    case 2: return a * b;            // EW_MUL
    case 3: return a / b;            // EW_DIV
    case 4: return fminf(a, b);      // EW_MIN
    default: return fmaxf(a, b);     // EW_MAX
  }
}

template <typename T, typename V8>
__global__ void ewise_transpose_k(T* __restrict__ out,
                                  const T* __restrict__ a,
                                  const T* __restrict__ b, int op,
                                  long rows, long cols) {
  __shared__ T tile[TT * TPAD];
  long tr = blockIdx.x / ((cols + TT - 1) / TT);
  long tc = blockIdx.x % ((cols + TT - 1) / TT);
  long r0 = tr * TT, c0 = tc * TT;

  const u32 tx = threadIdx.x & 7;
  const u32 ty = threadIdx.x >> 3;

#pragma unroll
  for (int h = 0; h < 2; ++h) {
    long r = r0 + ty + h * 32;
    long c = c0 + tx * 8;
    if (r < rows) {
      if (c + 8 <= cols) {
        V8 va = *(const V8*)(a + r * cols + c);
        V8 vb = *(const V8*)(b + r * cols + c);
        V8 vo;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          vo[j] = (T)ewt_apply(op, (float)va[j], (float)vb[j]);
        *(V8*)(tile + (ty + h * 32) * TPAD + tx * 8) = vo;
      } else {
        for (int j = 0; j < 8 && c + j < cols; ++j)
          tile[(ty + h * 32) * TPAD + tx * 8 + j] =
              (T)ewt_apply(op, (float)a[r * cols + c + j],
                           (float)b[r * cols + c + j]);
      }
    }
  }
  __syncthreads();

#pragma unroll
  for (int h = 0; h < 2; ++h) {
    long oc = c0 + ty + h * 32;
    long orow = r0 + tx * 8;
    if (oc < cols) {
      if (orow + 8 <= rows) {
        T v[8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          v[j] = tile[(tx * 8 + j) * TPAD + ty + h * 32];
        *(V8*)(out + oc * rows + orow) = *(const V8*)v;
      } else {
        for (int j = 0; j < 8 && orow + j < rows; ++j)
          out[oc * rows + orow + j] =
              tile[(tx * 8 + j) * TPAD + ty + h * 32];
      }
    }
  }
}

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

extern "C" void launch_ewise_transpose_bf16(int op, void* out, const void* a,
                                            const void* b, long rows,
                                            long cols, hipStream_t s) {
  long nt = ((rows + TT - 1) / TT) * ((cols + TT - 1) / TT);
  hipLaunchKernelGGL((ewise_transpose_k<bf16, bf16v8>), dim3((unsigned)nt),
                     dim3(256), 0, s, (bf16*)out, (const bf16*)a,
                     (const bf16*)b, op, rows, cols);
}

extern "C" void launch_ewise_transpose_f32(int op, float* out,
                                           const float* a, const float* b,
                                           long rows, long cols,
                                           hipStream_t s) {
  long nt = ((rows + TT - 1) / TT) * ((cols + TT - 1) / TT);
  hipLaunchKernelGGL((ewise_transpose_k<float, f32v8>), dim3((unsigned)nt),
                     dim3(256), 0, s, out, a, b, op, rows, cols);
}
