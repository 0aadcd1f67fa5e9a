// CSR x dense SpMM: C(f32, M x N) += A_csr(M x K) @ B(K x N).
//
// Reference equivalent: MatRel's per-block CSC sparse products
// (SURVEY.md §2.1 "Per-block math kernels": sparse x dense block products).
// Within-block sparsity complements the block-level sparsity of
// BlockMatrix (absent blocks); a block whose density is below
// MatrelConfig.sparse_threshold is stored CSR and multiplied here.
//
// Layout: one workgroup per (A row, N-chunk of 2048); each thread owns 8
// output columns (f32 accumulators in registers); the row's nonzeros are
// walked once, each contributing v * B[col, n0:n0+8] with 16-B vector
// loads of B (coalesced across the workgroup).
#include "common.h"

template <typename T, int V>
__global__ void spmm_csr_acc_k(float* __restrict__ C,
                               const int* __restrict__ crow,
                               const int* __restrict__ cols,
                               const T* __restrict__ vals,
                               const T* __restrict__ B, long M, long N) {
  typedef T vecT __attribute__((ext_vector_type(V)));
  const long row = blockIdx.x;
  const long nchunk = blockIdx.y;
  const long c0 = (nchunk * blockDim.x + threadIdx.x) * V;
  if (row >= M) return;
  const int beg = crow[row], end = crow[row + 1];
  if (beg == end) return;

  float acc[V];
#pragma unroll
  for (int j = 0; j < V; ++j) acc[j] = 0.0f;

  if (c0 + V <= N) {
    for (int p = beg; p < end; ++p) {
      const float v = (float)vals[p];
      const long bc = cols[p];
      vecT bv = *(const vecT*)(B + bc * N + c0);
#pragma unroll
      for (int j = 0; j < V; ++j) acc[j] += v * (float)bv[j];
    }
    float* cp = C + row * N + c0;
#pragma unroll
    for (int j = 0; j < V; ++j) cp[j] += acc[j];
  } else if (c0 < N) {
    for (int p = beg; p < end; ++p) {
      const float v = (float)vals[p];
      const long bc = cols[p];
      for (int j = 0; c0 + j < N; ++j)
        acc[j] += v * (float)B[bc * N + c0 + j];
    }
    float* cp = C + row * N + c0;
    for (int j = 0; c0 + j < N; ++j) cp[j] += acc[j];
  }
}

extern "C" void launch_spmm_csr_acc_bf16(float* C, const int* crow,
                                         const int* cols, const void* vals,
                                         const void* B, long M, long N,
                                         hipStream_t s) {
  dim3 grid((unsigned)M, (unsigned)((N + 2047) / 2048));
  hipLaunchKernelGGL((spmm_csr_acc_k<bf16, 8>), grid, dim3(256), 0, s, C,
                     crow, cols, (const bf16*)vals, (const bf16*)B, M, N);
}

extern "C" void launch_spmm_csr_acc_f32(float* C, const int* crow,
                                        const int* cols, const float* vals,
                                        const float* B, long M, long N,
                                        hipStream_t s) {
  dim3 grid((unsigned)M, (unsigned)((N + 1023) / 1024));
  hipLaunchKernelGGL((spmm_csr_acc_k<float, 4>), grid, dim3(256), 0, s, C,
                     crow, cols, vals, B, M, N);
}
