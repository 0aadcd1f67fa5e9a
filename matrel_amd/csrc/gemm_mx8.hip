// MX-fp8 (e4m3 + e8m0 block scales) GEMM for gfx950:
//   C(f32, MxN) += A8(MxK) @ Bt8(NxK)^T
// on v_mfma_scale_f32_32x32x64_f8f6f4 - the only path to the ~4.7 PF fp8
// rate on CDNA4 (cdna_hip_programming.md §3/§4: non-scaled fp8 runs at the
// bf16 rate; MX-scaled K=64/128 is the fast path).
//
// Operand semantics (verified empirically by scripts/mx_probe.hip on
// MI355X): per 64-lane wave, lane l carries 32 e4m3 bytes (v8i32) of
// row/col l&31, with the INTERLEAVED k map
//   hw_k(l, byte) = (l>>5)*16 + (byte & 15) + ((byte >> 4) & 1)*32,
// and ONE e8m0 scale byte (2^(x-127), byte0 of the i32, opsel 0) that
// covers hw block (l>>5) (= hw_k in [(l>>5)*32, +32)). To keep scale
// blocks CONTIGUOUS in original k (what csrc/mxconvert.hip produces),
// the fragment reads below place original k [kh*16, +16) in bytes 0-15
// and [32+kh*16, +16) in bytes 16-31 (kh = l>>5). C/D: the standard
// 32x32 map col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5).
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
      // bytes 0-15 <- orig k [kh*16, +16): 16-B slot index kh
      u32 byte = row * 64u + kh * 16u;
      byte ^= ((row >> 2) & 3) << 4;
      const char* p = (const char*)a_base + byte;
      af[m][0] = *(const int*)(p);
      af[m][1] = *(const int*)(p + 4);
      af[m][2] = *(const int*)(p + 8);
      af[m][3] = *(const int*)(p + 12);
      // bytes 16-31 <- orig k [32 + kh*16, +16): slot kh+2 = slot ^ 2;
      // the XOR swizzle commutes with the bit-5 flip
      const char* p2 = (const char*)a_base + (byte ^ 32u);
      af[m][4] = *(const int*)(p2);
      af[m][5] = *(const int*)(p2 + 4);
      af[m][6] = *(const int*)(p2 + 8);
      af[m][7] = *(const int*)(p2 + 12);
      sa[m] = Asc[(long)kblk * M + m0 + row];
    }
    v8i bf[2];
    int sb[2];
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      u32 row = brow + n * 32;
      u32 byte = row * 64u + kh * 16u;
      byte ^= ((row >> 2) & 3) << 4;
      const char* p = (const char*)b_base + byte;
      bf[n][0] = *(const int*)(p);
      bf[n][1] = *(const int*)(p + 4);
      bf[n][2] = *(const int*)(p + 8);
      bf[n][3] = *(const int*)(p + 12);
      const char* p2 = (const char*)b_base + (byte ^ 32u);
      bf[n][4] = *(const int*)(p2);
      bf[n][5] = *(const int*)(p2 + 4);
      bf[n][6] = *(const int*)(p2 + 8);
      bf[n][7] = *(const int*)(p2 + 12);
      sb[n] = Bsc[(long)kblk * N + n0 + row];
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
  const long ccol = n0 + wn * 64 + (lane & 31);
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

// ---- BK=128 variant: 2 scaled MFMAs per fragment per K-tile (amortizes
// the per-tile barrier/staging overhead that dominates at BK=64) and ONE
// dword scale load per fragment per tile (4 e8m0 bytes, extracted per
// kk/kh). LDS: 2 x (A 32 KiB + B 32 KiB) = 128 KiB.

__global__ void
__launch_bounds__(512)
gemm_bt_mx8_acc_k128(float* __restrict__ C,
                     const unsigned char* __restrict__ A,
                     const unsigned char* __restrict__ Asc,
                     const unsigned char* __restrict__ Bt,
                     const unsigned char* __restrict__ Bsc, int M, int N,
                     int K, int beta0) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  unsigned char* As = (unsigned char*)smem;  // [2][256*128]
  unsigned char* Bs = (unsigned char*)(smem + 2 * MXBM * 128);

  const u32 tid = threadIdx.x;
  const u32 lane = tid & 63, wave = tid >> 6;
  const u32 wm = wave >> 2, wn = wave & 3;

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
  const int KB = K / 32;

  // staging: tile = 256 rows x 128 B = 2048 chunks; 4 passes x 512 thr.
  // Swizzle for the 8-slot rows: slot ^ ((row>>1)&7).
  const u32 srow_b = tid >> 3;  // 0..63 (+p*64)
  const u32 sslot = tid & 7;

  auto stage = [&](int buf, int t) {
    const long k0 = (long)t * 128;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      u32 row = srow_b + p * 64;
      u32 ss = sslot ^ ((row >> 1) & 7);
      glds16(A + (m0 + row) * (long)K + k0 + ss * 16,
             As + buf * MXBM * 128 + row * 128u + sslot * 16u);
      glds16(Bt + (n0 + row) * (long)K + k0 + ss * 16,
             Bs + buf * MXBM * 128 + row * 128u + sslot * 16u);
    }
  };

  f32x16 acc[4][2] = {};
  const u32 arow = wm * 128 + (lane & 31);
  const u32 brow = wn * 64 + (lane & 31);
  const u32 kh = lane >> 5;

  // T5 static form: raise the second-dispatched half once (wave-uniform
  // condition via readfirstlane; see csrc/gemm.hip)
  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  stage(0, 0);
  __syncthreads();
  int cur = 0;
  const int NT = K / 128;
  for (int t = 0; t < NT; ++t) {
    if (t + 1 < NT) stage(cur ^ 1, t + 1);
    const unsigned char* a_base = As + cur * MXBM * 128;
    const unsigned char* b_base = Bs + cur * MXBM * 128;
    // one dword of scales per fragment row: blocks [4t, 4t+4)
    int sa_w[4], sb_w[2];  // byte b of the word = scale of block 4t+b
#pragma unroll
    for (int m = 0; m < 4; ++m) {
      long r = m0 + arow + m * 32;
      sa_w[m] = (int)Asc[(long)(t * 4 + 0) * M + r] |
                ((int)Asc[(long)(t * 4 + 1) * M + r] << 8) |
                ((int)Asc[(long)(t * 4 + 2) * M + r] << 16) |
                ((int)Asc[(long)(t * 4 + 3) * M + r] << 24);
    }
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      long r = n0 + brow + n * 32;
      sb_w[n] = (int)Bsc[(long)(t * 4 + 0) * N + r] |
                ((int)Bsc[(long)(t * 4 + 1) * N + r] << 8) |
                ((int)Bsc[(long)(t * 4 + 2) * N + r] << 16) |
                ((int)Bsc[(long)(t * 4 + 3) * N + r] << 24);
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      v8i af[4], bf[2];
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        u32 row = arow + m * 32;
        u32 byte = row * 128u + (u32)kk * 64u + kh * 16u;
        byte ^= ((row >> 1) & 7) << 4;
        const char* p = (const char*)a_base + byte;
        af[m][0] = *(const int*)(p);
        af[m][1] = *(const int*)(p + 4);
        af[m][2] = *(const int*)(p + 8);
        af[m][3] = *(const int*)(p + 12);
        const char* p2 = (const char*)a_base + (byte ^ 32u);
        af[m][4] = *(const int*)(p2);
        af[m][5] = *(const int*)(p2 + 4);
        af[m][6] = *(const int*)(p2 + 8);
        af[m][7] = *(const int*)(p2 + 12);
      }
#pragma unroll
      for (int n = 0; n < 2; ++n) {
        u32 row = brow + n * 32;
        u32 byte = row * 128u + (u32)kk * 64u + kh * 16u;
        byte ^= ((row >> 1) & 7) << 4;
        const char* p = (const char*)b_base + byte;
        bf[n][0] = *(const int*)(p);
        bf[n][1] = *(const int*)(p + 4);
        bf[n][2] = *(const int*)(p + 8);
        bf[n][3] = *(const int*)(p + 12);
        const char* p2 = (const char*)b_base + (byte ^ 32u);
        bf[n][4] = *(const int*)(p2);
        bf[n][5] = *(const int*)(p2 + 4);
        bf[n][6] = *(const int*)(p2 + 8);
        bf[n][7] = *(const int*)(p2 + 12);
      }
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 2; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              af[m], bf[n], acc[m][n], 0, 0, 0,
              (sa_w[m] >> (8 * (kk * 2 + (int)kh))) & 0xFF, 0,
              (sb_w[n] >> (8 * (kk * 2 + (int)kh))) & 0xFF);
    }
    __syncthreads();
    cur ^= 1;
  }

  const long ccol = n0 + wn * 64 + (lane & 31);
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
  if (K % 128 == 0) {
    size_t lds = 4u * MXBM * 128;  // 128 KiB
    hipLaunchKernelGGL(gemm_bt_mx8_acc_k128, grid, dim3(512), lds, stream,
                       C, (const unsigned char*)A,
                       (const unsigned char*)Asc, (const unsigned char*)Bt,
                       (const unsigned char*)Bsc, M, N, K, beta0);
    return;
  }
  size_t lds = 4u * MXBM * MXBK;  // 64 KiB
  hipLaunchKernelGGL(gemm_bt_mx8_acc_k, grid, dim3(512), lds, stream, C,
                     (const unsigned char*)A, (const unsigned char*)Asc,
                     (const unsigned char*)Bt, (const unsigned char*)Bsc, M,
                     N, K, beta0);
}
