// Fused elementwise / scalar / masking kernels (HBM-bandwidth-bound).
//
// Reference equivalent: the JVM elementwise passes of MatRel's LocalMatrix
// (SURVEY.md §2.2 row "fused HIP elementwise+transpose kernels"). Single
// pass, 16 B per lane (8 bf16 / 4 f32 - guideline 13: hipcc does not
// auto-vectorize bf16), grid-stride with a capped grid.
#include "common.h"
#include <math.h>

#define EW_ADD 0
#define EW_SUB 1
#define EW_MUL 2
#define EW_DIV 3
#define EW_MIN 4
#define EW_MAX 5
#define SC_ADD 0
#define SC_MUL 1
#define SC_POW 2

typedef short short8 __attribute__((ext_vector_type(8)));
typedef float float4v __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float ew_apply(int op, float a, float b) {
  switch (op) {
    case EW_ADD: return a + b;
    case EW_SUB: return a - b;
    case EW_MUL: return a * b;
    case EW_DIV: return a / b;
    case EW_MIN: return fminf(a, b);
    default:     return fmaxf(a, b);
  }
}

__device__ __forceinline__ float sc_apply(int op, float a, float s) {
  switch (op) {
    case SC_ADD: return a + s;
    case SC_MUL: return a * s;
    default:     return powf(a, s);
  }
}

// ---- bf16 binary elementwise: out = a OP b, n elements, n % 8 == 0 tail
// handled by a scalar remainder loop on the last block.
template <int OP>
__global__ void ewise_bf16_k(bf16* __restrict__ out,
                             const bf16* __restrict__ a,
                             const bf16* __restrict__ b, long n) {
  long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = i0; i + 8 <= n; i += stride) {
    bf16x8 va = *(const bf16x8*)(a + i);
    bf16x8 vb = *(const bf16x8*)(b + i);
    bf16x8 vo;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      vo[j] = (bf16)ew_apply(OP, (float)va[j], (float)vb[j]);
    *(bf16x8*)(out + i) = vo;
  }
  // tail
  long tail = n & 7;
  if (blockIdx.x == 0 && threadIdx.x < tail) {
    long i = n - tail + threadIdx.x;
    out[i] = (bf16)ew_apply(OP, (float)a[i], (float)b[i]);
  }
}

template <int OP>
__global__ void ewise_f32_k(float* __restrict__ out,
                            const float* __restrict__ a,
                            const float* __restrict__ b, long n) {
  long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long stride = (long)gridDim.x * blockDim.x * 4;
  for (long i = i0; i + 4 <= n; i += stride) {
    float4v va = *(const float4v*)(a + i);
    float4v vb = *(const float4v*)(b + i);
    float4v vo;
#pragma unroll
    for (int j = 0; j < 4; ++j) vo[j] = ew_apply(OP, va[j], vb[j]);
    *(float4v*)(out + i) = vo;
  }
  long tail = n & 3;
  if (blockIdx.x == 0 && threadIdx.x < tail) {
    long i = n - tail + threadIdx.x;
    out[i] = ew_apply(OP, a[i], b[i]);
  }
}

// ---- scalar ops
template <int OP>
__global__ void ewise_scalar_bf16_k(bf16* __restrict__ out,
                                    const bf16* __restrict__ a, float s,
                                    long n) {
  long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = i0; i + 8 <= n; i += stride) {
    bf16x8 va = *(const bf16x8*)(a + i);
    bf16x8 vo;
#pragma unroll
    for (int j = 0; j < 8; ++j) vo[j] = (bf16)sc_apply(OP, (float)va[j], s);
    *(bf16x8*)(out + i) = vo;
  }
  long tail = n & 7;
  if (blockIdx.x == 0 && threadIdx.x < tail) {
    long i = n - tail + threadIdx.x;
    out[i] = (bf16)sc_apply(OP, (float)a[i], s);
  }
}

template <int OP>
__global__ void ewise_scalar_f32_k(float* __restrict__ out,
                                   const float* __restrict__ a, float s,
                                   long n) {
  long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long stride = (long)gridDim.x * blockDim.x * 4;
  for (long i = i0; i + 4 <= n; i += stride) {
    float4v va = *(const float4v*)(a + i);
    float4v vo;
#pragma unroll
    for (int j = 0; j < 4; ++j) vo[j] = sc_apply(OP, va[j], s);
    *(float4v*)(out + i) = vo;
  }
  long tail = n & 3;
  if (blockIdx.x == 0 && threadIdx.x < tail) {
    long i = n - tail + threadIdx.x;
    out[i] = sc_apply(OP, a[i], s);
  }
}

// ---- value mask: keep values in [lo, hi] (or outside), zero the rest.
template <typename T>
__global__ void mask_value_k(T* __restrict__ out, const T* __restrict__ a,
                             float lo, float hi, int keep_inside, long n) {
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = i0; i < n; i += stride) {
    float v = (float)a[i];
    bool inside = (v >= lo) && (v <= hi);
    bool keep = keep_inside ? inside : !inside;
    out[i] = keep ? a[i] : (T)0.0f;
  }
}

// ---- row/col index mask: out[r][c] = a[r][c] * rm[r] * cm[c]
template <typename T>
__global__ void mask_rows_cols_k(T* __restrict__ out,
                                 const T* __restrict__ a,
                                 const unsigned char* rm,
                                 const unsigned char* cm, long rows,
                                 long cols) {
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long n = rows * cols;
  for (long i = i0; i < n; i += stride) {
    long r = i / cols, c = i - r * cols;
    bool keep = (!rm || rm[r]) && (!cm || cm[c]);
    out[i] = keep ? a[i] : (T)0.0f;
  }
}

// ---- outer "cross join": out[i][j] = u[i] OP v[j] (f32 vectors)
template <int OP>
__global__ void outer_k(float* __restrict__ out, const float* __restrict__ u,
                        const float* __restrict__ v, long m, long n) {
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long total = m * n;
  for (long i = i0; i < total; i += stride) {
    long r = i / n, c = i - r * n;
    out[i] = ew_apply(OP == 0 ? EW_MUL : EW_ADD, u[r], v[c]);
  }
}

// ---- value semi-join: keep a[i] iff its value appears in the SORTED
// needle array (binary search per element); else zero. Used by the
// relational value-join (SURVEY.md §2.1 "joins on matrix indices/values").
template <typename T>
__global__ void mask_isin_k(T* __restrict__ out, const T* __restrict__ a,
                            const float* __restrict__ needles, int nn,
                            long n) {
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = i0; i < n; i += stride) {
    float v = (float)a[i];
    int lo = 0, hi = nn - 1;
    bool found = false;
    while (lo <= hi) {
      int mid = (lo + hi) >> 1;
      float m = needles[mid];
      if (m == v) { found = true; break; }
      if (m < v) lo = mid + 1; else hi = mid - 1;
    }
    out[i] = found ? a[i] : (T)0.0f;
  }
}

// ------------------------------------------------------------- launchers --

static inline int ew_grid(long n, int per_thread) {
  long blocks = (n + 256L * per_thread - 1) / (256L * per_thread);
  return (int)(blocks < 2048 ? (blocks < 1 ? 1 : blocks) : 2048);
}

extern "C" void launch_ewise_bf16(int op, void* out, const void* a,
                                  const void* b, long n, hipStream_t s) {
  dim3 g(ew_grid(n, 8)), blk(256);
  switch (op) {
    case 0: hipLaunchKernelGGL(ewise_bf16_k<0>, g, blk, 0, s, (bf16*)out, (const bf16*)a, (const bf16*)b, n); break;
    case 1: hipLaunchKernelGGL(ewise_bf16_k<1>, g, blk, 0, s, (bf16*)out, (const bf16*)a, (const bf16*)b, n); break;
    case 2: hipLaunchKernelGGL(ewise_bf16_k<2>, g, blk, 0, s, (bf16*)out, (const bf16*)a, (const bf16*)b, n); break;
    case 3: hipLaunchKernelGGL(ewise_bf16_k<3>, g, blk, 0, s, (bf16*)out, (const bf16*)a, (const bf16*)b, n); break;
    case 4: hipLaunchKernelGGL(ewise_bf16_k<4>, g, blk, 0, s, (bf16*)out, (const bf16*)a, (const bf16*)b, n); break;
    default: hipLaunchKernelGGL(ewise_bf16_k<5>, g, blk, 0, s, (bf16*)out, (const bf16*)a, (const bf16*)b, n); break;
  }
}

extern "C" void launch_ewise_f32(int op, float* out, const float* a,
                                 const float* b, long n, hipStream_t s) {
  dim3 g(ew_grid(n, 4)), blk(256);
  switch (op) {
    case 0: hipLaunchKernelGGL(ewise_f32_k<0>, g, blk, 0, s, out, a, b, n); break;
    case 1: hipLaunchKernelGGL(ewise_f32_k<1>, g, blk, 0, s, out, a, b, n); break;
    case 2: hipLaunchKernelGGL(ewise_f32_k<2>, g, blk, 0, s, out, a, b, n); break;
    case 3: hipLaunchKernelGGL(ewise_f32_k<3>, g, blk, 0, s, out, a, b, n); break;
    case 4: hipLaunchKernelGGL(ewise_f32_k<4>, g, blk, 0, s, out, a, b, n); break;
    default: hipLaunchKernelGGL(ewise_f32_k<5>, g, blk, 0, s, out, a, b, n); break;
  }
}

extern "C" void launch_ewise_scalar_bf16(int op, void* out, const void* a,
                                         float sc, long n, hipStream_t s) {
  dim3 g(ew_grid(n, 8)), blk(256);
  switch (op) {
    case 0: hipLaunchKernelGGL(ewise_scalar_bf16_k<0>, g, blk, 0, s, (bf16*)out, (const bf16*)a, sc, n); break;
    case 1: hipLaunchKernelGGL(ewise_scalar_bf16_k<1>, g, blk, 0, s, (bf16*)out, (const bf16*)a, sc, n); break;
    default: hipLaunchKernelGGL(ewise_scalar_bf16_k<2>, g, blk, 0, s, (bf16*)out, (const bf16*)a, sc, n); break;
  }
}

extern "C" void launch_ewise_scalar_f32(int op, float* out, const float* a,
                                        float sc, long n, hipStream_t s) {
  dim3 g(ew_grid(n, 4)), blk(256);
  switch (op) {
    case 0: hipLaunchKernelGGL(ewise_scalar_f32_k<0>, g, blk, 0, s, out, a, sc, n); break;
    case 1: hipLaunchKernelGGL(ewise_scalar_f32_k<1>, g, blk, 0, s, out, a, sc, n); break;
    default: hipLaunchKernelGGL(ewise_scalar_f32_k<2>, g, blk, 0, s, out, a, sc, n); break;
  }
}

extern "C" void launch_mask_value_bf16(void* out, const void* a, float lo,
                                       float hi, int inside, long n,
                                       hipStream_t s) {
  hipLaunchKernelGGL(mask_value_k<bf16>, dim3(ew_grid(n, 1)), dim3(256), 0, s,
                     (bf16*)out, (const bf16*)a, lo, hi, inside, n);
}

extern "C" void launch_mask_value_f32(float* out, const float* a, float lo,
                                      float hi, int inside, long n,
                                      hipStream_t s) {
  hipLaunchKernelGGL(mask_value_k<float>, dim3(ew_grid(n, 1)), dim3(256), 0,
                     s, out, a, lo, hi, inside, n);
}

extern "C" void launch_mask_rows_cols_bf16(void* out, const void* a,
                                           const unsigned char* rm,
                                           const unsigned char* cm, long rows,
                                           long cols, hipStream_t s) {
  hipLaunchKernelGGL(mask_rows_cols_k<bf16>, dim3(ew_grid(rows * cols, 1)),
                     dim3(256), 0, s, (bf16*)out, (const bf16*)a, rm, cm,
                     rows, cols);
}

extern "C" void launch_mask_rows_cols_f32(float* out, const float* a,
                                          const unsigned char* rm,
                                          const unsigned char* cm, long rows,
                                          long cols, hipStream_t s) {
  hipLaunchKernelGGL(mask_rows_cols_k<float>, dim3(ew_grid(rows * cols, 1)),
                     dim3(256), 0, s, out, a, rm, cm, rows, cols);
}

extern "C" void launch_outer_f32(int op, float* out, const float* u,
                                 const float* v, long m, long n,
                                 hipStream_t s) {
  dim3 g(ew_grid(m * n, 1)), blk(256);
  if (op == 0)
    hipLaunchKernelGGL(outer_k<0>, g, blk, 0, s, out, u, v, m, n);
  else
    hipLaunchKernelGGL(outer_k<1>, g, blk, 0, s, out, u, v, m, n);
}

extern "C" void launch_mask_isin_bf16(void* out, const void* a,
                                      const float* needles, int nn, long n,
                                      hipStream_t s) {
  hipLaunchKernelGGL(mask_isin_k<bf16>, dim3(ew_grid(n, 1)), dim3(256), 0, s,
                     (bf16*)out, (const bf16*)a, needles, nn, n);
}

extern "C" void launch_mask_isin_f32(float* out, const float* a,
                                     const float* needles, int nn, long n,
                                     hipStream_t s) {
  hipLaunchKernelGGL(mask_isin_k<float>, dim3(ew_grid(n, 1)), dim3(256), 0,
                     s, out, a, needles, nn, n);
}
