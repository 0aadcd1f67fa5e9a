// Raw v_mfma_scale_f32_32x32x64_f8f6f4 issue-rate probe per operand
// format (cbsz/blgp = 0:fp8, 2:fp6, 4:fp4). Operands live in registers
// (no LDS, no memory in the loop) - measures the pure pipe rate the
// silicon gives each format, i.e. the ceiling for the mx8/mx6/mx4 GEMM
// kernels. 8 independent acc chains/wave hide the 2-wave/SIMD sharing.
// Build: hipcc --offload-arch=gfx950 -O3 scripts/mfma_rate.hip -o scripts/mfma_rate
#include <hip/hip_runtime.h>
#include <cstdio>

typedef int v8i __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

template <int FMT>
__global__ void __launch_bounds__(512) rate_k(float* out, int iters) {
  v8i a = {}, b = {};
  a[0] = (int)(threadIdx.x * 2654435761u);
  b[0] = (int)(threadIdx.x * 40503u + 7);
  f32x16 acc[8] = {};
  for (int i = 0; i < iters; ++i) {
#pragma unroll
    for (int c = 0; c < 8; ++c)
      acc[c] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
          a, b, acc[c], FMT, FMT, 0, 127, 0, 127);
  }
  float s = 0;
#pragma unroll
  for (int c = 0; c < 8; ++c) s += acc[c][0];
  if (s == 1234.5f) out[threadIdx.x] = s;  // never true; keeps the loop
}

template <int FMT>
static void run(const char* name, int iters) {
  float* d;
  (void)hipMalloc(&d, 4096);
  dim3 grid(512), blk(512);
  hipLaunchKernelGGL(rate_k<FMT>, grid, blk, 0, 0, d, 16);  // warm
  (void)hipDeviceSynchronize();
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);
  (void)hipEventRecord(e0);
  hipLaunchKernelGGL(rate_k<FMT>, grid, blk, 0, 0, d, iters);
  (void)hipEventRecord(e1);
  (void)hipEventSynchronize(e1);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, e0, e1);
  // instrs = grid*blk/64 waves * iters * 8; flops = instr * 2*32*32*64
  double waves = 512.0 * 512.0 / 64.0;
  double fl = waves * (double)iters * 8.0 * 131072.0;
  printf("%s: %.0f TF (%.2f ms)\n", name, fl / (ms * 1e-3) / 1e12, ms);
  (void)hipFree(d);
}

int main() {
  int it = 20000;
  for (int rep = 0; rep < 2; ++rep) {
    run<0>("fp8 (cbsz=0)", it);
    run<2>("fp6 (cbsz=2)", it);
    run<4>("fp4 (cbsz=4)", it);
  }
  return 0;
}
