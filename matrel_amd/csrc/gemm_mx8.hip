// MX-fp8 (e4m3 + e8m0 block scales) GEMM for gfx950:
//   C(f32, MxN) += A8(MxK) @ Bt8(NxK)^T
// on v_mfma_scale_f32_32x32x64_f8f6f4 - the only path to the ~4.7 PF fp8
// rate on CDNA4 (cdna_hip_programming.md §3/§4: non-scaled fp8 runs at the
// bf16 rate; MX-scaled K=64/128 is the fast path).
//
// Operand semantics (verified empirically by scripts/mx_probe.hip on
// MI355X): per 64-lane wave, lane l carries
//   A: row = l&31, k = (l>>5)*32 + byte  (32 e4m3 bytes = v8i32)
//   B: col = l&31, same k packing
//   scale: ONE e8m0 byte (2^(x-127)) in byte0 of the i32 operand (opsel
//   0), covering the lane's 32-wide k-block; C/D: the standard 32x32 map
//   col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5).
//
// Storage: A8/Bt8 row-major uint8 (e4m3 bits); scales row-major uint8
// [M][K/32] (e8m0), produced by csrc/mxconvert.hip.
//
// Structure: 256x256 tile, 8 waves (2Mx4N), BK=64 (one scaled MFMA per
// 32x32 fragment per K-tile), global_load_lds staging with an XOR slot
// swizzle on the 64-B fp8 rows, double-buffered LDS.
#include "common.h"

typedef int v8i __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

#define MXBM 256
#define MXBN 256
#define MXBK 64

__global__ void
__launch_bounds__(512)
gemm_bt_mx8_acc_k(float* __restrict__ C, const unsigned char* __restrict__ A,
                  const unsigned char* __restrict__ Asc,
                  const unsigned char* __restrict__ Bt,
                  const unsigned char* __restrict__ Bsc, int M, int N,
                  int K, int beta0) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // [2 buf][A tile 256x64B | B tile 256x64B] = 2 x 32 KiB
  unsigned char* As = (unsigned char*)smem;                   // [2][256*64]
  unsigned char* Bs = (unsigned char*)(smem + 2 * MXBM * MXBK);

  const u32 tid = threadIdx.x;
  const u32 lane = tid & 63, wave = tid >> 6;
  const u32 wm = wave >> 2, wn = wave & 3;  // 2 x 4 wave grid; 128x64 out

  const u32 ntm = (u32)(M / MXBM), ntn = (u32)(N / MXBN);
  u32 bid = xcd_swizzle(blockIdx.x, ntm * ntn);
  u32 tm, tn;
  if (ntm % 4 == 0 && ntn % 4 == 0) {
    supertile_map(bid, ntm, ntn, 4, 4, &tm, &tn);
  } else {
    tm = bid / ntn;
    tn = bid % ntn;
  }
  const long m0 = (long)tm * MXBM, n0 = (long)tn * MXBN;
  const int KB = K / 32;  // scale blocks per row

  // staging: operand tile = 256 rows x 64 B = 1024 x 16-B chunks;
  // 512 threads x 2 glds per operand. Source slot pre-swizzled so the
  // linear LDS image equals the swizzled layout the reads expect
  // (slot ^ ((row>>2)&3) - conflict-free for the 32-B fragment reads).
  const u32 srow = tid >> 2;        // 0..127 (row per pass; 2 passes)
  const u32 sslot = tid & 3;        // 4 x 16-B slots per 64-B row
  const u32 src_slot = sslot ^ ((srow >> 2) & 3);

  auto stage = [&](int buf, int t) {
    const long k0 = (long)t * MXBK;
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      u32 row = srow + half * 128;
      glds16(A + (m0 + row) * (long)K + k0 + src_slot * 16,
             As + buf * MXBM * MXBK + row * 64u + sslot * 16u);
      glds16(Bt + (n0 + row) * (long)K + k0 + src_slot * 16,
             Bs + buf * MXBM * MXBK + row * 64u + sslot * 16u);
    }
  };

  f32x16 acc[4][2] = {};  // [m-frag 32][n-frag 32]
  const u32 arow = wm * 128 + (lane & 31);  // + m*32
  const u32 brow = wn * 64 + (lane & 31);   // + n*32
  const u32 kh = lane >> 5;                 // k half (32-block) of the tile

  stage(0, 0);
  __syncthreads();
  int cur = 0;
  const int NT = K / MXBK;
  for (int t = 0; t < NT; ++t) {
    if (t + 1 < NT) stage(cur ^ 1, t + 1);
    const unsigned char* a_base = As + cur * MXBM * MXBK;
    const unsigned char* b_base = Bs + cur * MXBM * MXBK;
    const int kblk = t * 2 + (int)kh;
    // fragment + scale loads
    v8i af[4];
    int sa[4];
#pragma unroll
    for (int m = 0; m < 4; ++m) {
      u32 row = arow + m * 32;
      u32 byte = row * 64u + kh * 32u;
      byte ^= ((row >> 2) & 3) << 4;
      const char* p = (const char*)a_base + byte;
      af[m][0] = *(const int*)(p);
      af[m][1] = *(const int*)(p + 4);
      af[m][2] = *(const int*)(p + 8);
      af[m][3] = *(const int*)(p + 12);
      // second 16-B slot of the 32-B fragment: slot low-bit flip is
      // preserved by the XOR swizzle (s^sw and (s+1)^sw differ in bit 0)
      const char* p2 = (const char*)a_base + (byte ^ 16u);
      af[m][4] = *(const int*)(p2);
      af[m][5] = *(const int*)(p2 + 4);
      af[m][6] = *(const int*)(p2 + 8);
      af[m][7] = *(const int*)(p2 + 12);
      sa[m] = Asc[(m0 + row) * (long)KB + kblk];
    }
    v8i bf[2];
    int sb[2];
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      u32 row = brow + n * 32;
      u32 byte = row * 64u + kh * 32u;
      byte ^= ((row >> 2) & 3) << 4;
      const char* p = (const char*)b_base + byte;
      bf[n][0] = *(const int*)(p);
      bf[n][1] = *(const int*)(p + 4);
      bf[n][2] = *(const int*)(p + 8);
      bf[n][3] = *(const int*)(p + 12);
      const char* p2 = (const char*)b_base + (byte ^ 16u);
      bf[n][4] = *(const int*)(p2);
      bf[n][5] = *(const int*)(p2 + 4);
      bf[n][6] = *(const int*)(p2 + 8);
      bf[n][7] = *(const int*)(p2 + 12);
      sb[n] = Bsc[(n0 + row) * (long)KB + kblk];
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < 2; ++n)
        acc[m][n] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            af[m], bf[n], acc[m][n], 0, 0, 0, sa[m], 0, sb[n]);
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
    cur ^= 1;
  }

  // epilogue: standard 32x32 C/D map, batched loads
  const u32 ccol = wn * 64 + (lane & 31);
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      float cv[16];
      if (!beta0) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          long row = m0 + arow + m * 32 - (lane & 31) + (r & 3) +
                     8 * (r >> 2) + 4 * (lane >> 5);
          cv[r] = C[row * (long)N + ccol + n * 32];
        }
      }
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        long row = m0 + arow + m * 32 - (lane & 31) + (r & 3) +
                   8 * (r >> 2) + 4 * (lane >> 5);
        C[row * (long)N + ccol + n * 32] =
            (beta0 ? 0.0f : cv[r]) + acc[m][n][r];
      }
    }
}

extern "C" void launch_gemm_bt_mx8_acc(float* C, const void* A,
                                       const void* Asc, const void* Bt,
                                       const void* Bsc, int M, int N, int K,
                                       int beta0, hipStream_t stream) {
  dim3 grid((M / MXBM) * (N / MXBN));
  size_t lds = 4u * MXBM * MXBK;  // 64 KiB
  hipLaunchKernelGGL(gemm_bt_mx8_acc_k, grid, dim3(512), lds, stream, C,
                     (const unsigned char*)A, (const unsigned char*)Asc,
                     (const unsigned char*)Bt, (const unsigned char*)Bsc, M,
                     N, K, beta0);
}
