// Reduction kernels: full / per-row / per-col {sum, max, min, nnz, sqsum}
// + trace. fp32 accumulation throughout.
//
// Reference equivalent: MatRel AggregateOperator executions (sum / rowSum /
// colSum / count / max / min / trace, SURVEY.md §2.1 "Physical relational
// ops"). Full reductions run as grid partials + a one-block combine
// (deterministic - no float atomics).
#include "common.h"
#include <float.h>

#define RK_SUM 0
#define RK_MAX 1
#define RK_MIN 2
#define RK_NNZ 3
#define RK_SQSUM 4

__device__ __forceinline__ float rk_init(int kind) {
  switch (kind) {
    case RK_MAX: return -FLT_MAX;
    case RK_MIN: return FLT_MAX;
    default:     return 0.0f;
  }
}

__device__ __forceinline__ float rk_map(int kind, float v) {
  switch (kind) {
    case RK_NNZ:   return v != 0.0f ? 1.0f : 0.0f;
    case RK_SQSUM: return v * v;
    default:       return v;
  }
}

__device__ __forceinline__ float rk_comb(int kind, float a, float b) {
  switch (kind) {
    case RK_MAX: return fmaxf(a, b);
    case RK_MIN: return fminf(a, b);
    default:     return a + b;
  }
}

template <typename T>
__device__ __forceinline__ float load_f(const T* p, long i) {
  return (float)p[i];
}

// ---- block-level reduce of a thread-local value -> lane0 of wave 0
__device__ __forceinline__ float block_reduce(int kind, float v) {
  __shared__ float warp_part[4];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = rk_comb(kind, v, __shfl_down(v, off, 64));
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) warp_part[wave] = v;
  __syncthreads();
  if (wave == 0) {
    v = lane < 4 ? warp_part[lane] : rk_init(kind);
#pragma unroll
    for (int off = 2; off > 0; off >>= 1)
      v = rk_comb(kind, v, __shfl_down(v, off, 64));
  }
  return v;
}

// ---- full reduction: stage 1 (per-block partials), 16-B vector loads
template <typename T, int KIND, int V>
__global__ void reduce_full_stage1_k(float* __restrict__ partials,
                                     const T* __restrict__ a, long n) {
  typedef T vecT __attribute__((ext_vector_type(V)));
  float acc = rk_init(KIND);
  long nv = n / V;
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = i0; i < nv; i += stride) {
    vecT v = *(const vecT*)(a + i * V);
#pragma unroll
    for (int j = 0; j < V; ++j)
      acc = rk_comb(KIND, acc, rk_map(KIND, (float)v[j]));
  }
  if (blockIdx.x == 0 && threadIdx.x < (n - nv * V)) {
    long i = nv * V + threadIdx.x;
    acc = rk_comb(KIND, acc, rk_map(KIND, load_f(a, i)));
  }
  float r = block_reduce(KIND, acc);
  if (threadIdx.x == 0) partials[blockIdx.x] = r;
}

// ---- stage 2: one block combines partials
__global__ void reduce_full_stage2_k(float* __restrict__ out,
                                     const float* __restrict__ partials,
                                     int nparts, int kind) {
  float acc = rk_init(kind);
  for (int i = threadIdx.x; i < nparts; i += blockDim.x)
    acc = rk_comb(kind, acc, partials[i]);
  float r = block_reduce(kind, acc);
  if (threadIdx.x == 0) out[0] = r;
}

// ---- per-row reduction: one block per row (out[m]), 16-B vector loads
template <typename T, int KIND, int V>
__global__ void reduce_rows_k(float* __restrict__ out,
                              const T* __restrict__ a, long rows,
                              long cols) {
  typedef T vecT __attribute__((ext_vector_type(V)));
  long r = blockIdx.x;
  if (r >= rows) return;
  const T* row = a + r * cols;
  float acc = rk_init(KIND);
  long cv = cols / V;
  for (long c = threadIdx.x; c < cv; c += blockDim.x) {
    vecT v = *(const vecT*)(row + c * V);
#pragma unroll
    for (int j = 0; j < V; ++j)
      acc = rk_comb(KIND, acc, rk_map(KIND, (float)v[j]));
  }
  if (threadIdx.x < (cols - cv * V))
    acc = rk_comb(KIND, acc, rk_map(KIND, load_f(row, cv * V + threadIdx.x)));
  float v = block_reduce(KIND, acc);
  if (threadIdx.x == 0) out[r] = v;
}

// ---- per-col reduction, two-stage: grid (col-tiles x row-chunks), each
// block reduces its row chunk for 2048 columns (8 cols/thread, 16-B
// vector loads) into partials[chunk][cols]; stage 2 combines chunks.
#define CR_VEC 8  // bf16 elems per load; f32 uses 4

template <typename T, int KIND, int V>
__global__ void reduce_cols_stage1_k(float* __restrict__ partials,
                                     const T* __restrict__ a, long rows,
                                     long cols, int nchunks) {
  typedef T vecT __attribute__((ext_vector_type(V)));
  long c0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * V;
  int chunk = blockIdx.y;
  long r0 = rows * chunk / nchunks, r1 = rows * (chunk + 1) / nchunks;
  float acc[V];
#pragma unroll
  for (int j = 0; j < V; ++j) acc[j] = rk_init(KIND);
  if (c0 + V <= cols) {
    // 4-deep row unroll: keeps 4 independent 16-B loads in flight per
    // thread (single-load chains leave the column reduction latency-
    // bound at ~half of HBM bandwidth)
    long r = r0;
    for (; r + 4 <= r1; r += 4) {
      vecT v0 = *(const vecT*)(a + (r + 0) * cols + c0);
      vecT v1 = *(const vecT*)(a + (r + 1) * cols + c0);
      vecT v2 = *(const vecT*)(a + (r + 2) * cols + c0);
      vecT v3 = *(const vecT*)(a + (r + 3) * cols + c0);
#pragma unroll
      for (int j = 0; j < V; ++j) {
        acc[j] = rk_comb(KIND, acc[j], rk_map(KIND, (float)v0[j]));
        acc[j] = rk_comb(KIND, acc[j], rk_map(KIND, (float)v1[j]));
        acc[j] = rk_comb(KIND, acc[j], rk_map(KIND, (float)v2[j]));
        acc[j] = rk_comb(KIND, acc[j], rk_map(KIND, (float)v3[j]));
      }
    }
    for (; r < r1; ++r) {
      vecT v = *(const vecT*)(a + r * cols + c0);
#pragma unroll
      for (int j = 0; j < V; ++j)
        acc[j] = rk_comb(KIND, acc[j], rk_map(KIND, (float)v[j]));
    }
  } else if (c0 < cols) {
    for (long r = r0; r < r1; ++r)
      for (int j = 0; c0 + j < cols; ++j)
        acc[j] = rk_comb(KIND, acc[j],
                         rk_map(KIND, load_f(a, r * cols + c0 + j)));
  } else {
    return;
  }
  float* p = partials + (long)chunk * cols + c0;
  for (int j = 0; j < V && c0 + j < cols; ++j) p[j] = acc[j];
}

__global__ void reduce_cols_stage2_k(float* __restrict__ out,
                                     const float* __restrict__ partials,
                                     long cols, int nchunks, int kind) {
  long c = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= cols) return;
  float acc = rk_init(kind);
  for (int ch = 0; ch < nchunks; ++ch)
    acc = rk_comb(kind, acc, partials[(long)ch * cols + c]);
  out[c] = acc;
}

// ---- trace: sum of diagonal
template <typename T>
__global__ void trace_stage1_k(float* __restrict__ partials,
                               const T* __restrict__ a, long n, long cols) {
  float acc = 0.0f;
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = i0; i < n; i += stride) acc += load_f(a, i * cols + i);
  float r = block_reduce(RK_SUM, acc);
  if (threadIdx.x == 0) partials[blockIdx.x] = r;
}

// ------------------------------------------------------------- launchers --

static inline int rgrid(long n) {
  long b = (n + 255) / 256;
  return (int)(b < 1 ? 1 : (b > 1024 ? 1024 : b));
}

#define K_DISPATCH(T, V, fn, ...)                                   \
  switch (kind) {                                                   \
    case 0: hipLaunchKernelGGL((fn<T, 0, V>), __VA_ARGS__); break;  \
    case 1: hipLaunchKernelGGL((fn<T, 1, V>), __VA_ARGS__); break;  \
    case 2: hipLaunchKernelGGL((fn<T, 2, V>), __VA_ARGS__); break;  \
    case 3: hipLaunchKernelGGL((fn<T, 3, V>), __VA_ARGS__); break;  \
    default: hipLaunchKernelGGL((fn<T, 4, V>), __VA_ARGS__); break; \
  }

extern "C" void launch_reduce_full_bf16(float* out, float* partials,
                                        const void* a, long n, int kind,
                                        hipStream_t s) {
  int g = rgrid(n);
  K_DISPATCH(bf16, 8, reduce_full_stage1_k, dim3(g), dim3(256), 0, s, partials,
             (const bf16*)a, n);
  hipLaunchKernelGGL(reduce_full_stage2_k, dim3(1), dim3(256), 0, s, out,
                     partials, g, kind);
}

extern "C" void launch_reduce_full_f32(float* out, float* partials,
                                       const float* a, long n, int kind,
                                       hipStream_t s) {
  int g = rgrid(n);
  K_DISPATCH(float, 4, reduce_full_stage1_k, dim3(g), dim3(256), 0, s, partials,
             a, n);
  hipLaunchKernelGGL(reduce_full_stage2_k, dim3(1), dim3(256), 0, s, out,
                     partials, g, kind);
}

extern "C" void launch_reduce_rows_bf16(float* out, const void* a, long rows,
                                        long cols, int kind, hipStream_t s) {
  K_DISPATCH(bf16, 8, reduce_rows_k, dim3((unsigned)rows), dim3(256), 0, s, out,
             (const bf16*)a, rows, cols);
}

extern "C" void launch_reduce_rows_f32(float* out, const float* a, long rows,
                                       long cols, int kind, hipStream_t s) {
  K_DISPATCH(float, 4, reduce_rows_k, dim3((unsigned)rows), dim3(256), 0, s,
             out, a, rows, cols);
}

// partials workspace supplied by the binding: [nchunks][cols] f32.
static inline int col_chunks(long rows, long cols) {
  // enough blocks to fill 256 CUs: col_tiles * chunks >= ~1024
  long col_tiles = (cols + 2047) / 2048;
  long want = (2048 + col_tiles - 1) / col_tiles;
  long mx = rows < 64 ? rows : 64;
  long c = want < 1 ? 1 : (want > mx ? mx : want);
  return (int)(c < 1 ? 1 : c);
}

#define KC_DISPATCH(T, V, ...)                                              \
  switch (kind) {                                                           \
    case 0: hipLaunchKernelGGL((reduce_cols_stage1_k<T, 0, V>), __VA_ARGS__); break; \
    case 1: hipLaunchKernelGGL((reduce_cols_stage1_k<T, 1, V>), __VA_ARGS__); break; \
    case 2: hipLaunchKernelGGL((reduce_cols_stage1_k<T, 2, V>), __VA_ARGS__); break; \
    case 3: hipLaunchKernelGGL((reduce_cols_stage1_k<T, 3, V>), __VA_ARGS__); break; \
    default: hipLaunchKernelGGL((reduce_cols_stage1_k<T, 4, V>), __VA_ARGS__); break; \
  }

extern "C" int reduce_cols_nchunks(long rows, long cols) {
  return col_chunks(rows, cols);
}

extern "C" void launch_reduce_cols_bf16(float* out, float* partials,
                                        const void* a, long rows, long cols,
                                        int kind, hipStream_t s) {
  int nc = col_chunks(rows, cols);
  dim3 g1((unsigned)((cols + 2047) / 2048), (unsigned)nc);
  KC_DISPATCH(bf16, 8, g1, dim3(256), 0, s, partials, (const bf16*)a, rows,
              cols, nc);
  hipLaunchKernelGGL(reduce_cols_stage2_k,
                     dim3((unsigned)((cols + 255) / 256)), dim3(256), 0, s,
                     out, partials, cols, nc, kind);
}

extern "C" void launch_reduce_cols_f32(float* out, float* partials,
                                       const float* a, long rows, long cols,
                                       int kind, hipStream_t s) {
  int nc = col_chunks(rows, cols);
  dim3 g1((unsigned)((cols + 1023) / 1024), (unsigned)nc);
  KC_DISPATCH(float, 4, g1, dim3(256), 0, s, partials, a, rows, cols, nc);
  hipLaunchKernelGGL(reduce_cols_stage2_k,
                     dim3((unsigned)((cols + 255) / 256)), dim3(256), 0, s,
                     out, partials, cols, nc, kind);
}

extern "C" void launch_trace_bf16(float* out, float* partials, const void* a,
                                  long n, long cols, hipStream_t s) {
  int g = rgrid(n);
  hipLaunchKernelGGL(trace_stage1_k<bf16>, dim3(g), dim3(256), 0, s,
                     partials, (const bf16*)a, n, cols);
  hipLaunchKernelGGL(reduce_full_stage2_k, dim3(1), dim3(256), 0, s, out,
                     partials, g, RK_SUM);
}

extern "C" void launch_trace_f32(float* out, float* partials, const float* a,
                                 long n, long cols, hipStream_t s) {
  int g = rgrid(n);
  hipLaunchKernelGGL(trace_stage1_k<float>, dim3(g), dim3(256), 0, s,
                     partials, a, n, cols);
  hipLaunchKernelGGL(reduce_full_stage2_k, dim3(1), dim3(256), 0, s, out,
                     partials, g, RK_SUM);
}
