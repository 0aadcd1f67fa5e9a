// MX-fp4 (e2m1 + e8m0 block scales) GEMM for gfx950:
//   C(f32, MxN) += A4(MxK) @ Bt4(NxK)^T
// on v_mfma_scale_f32_32x32x64_f8f6f4 with cbsz=blgp=4 - CDNA4's ~10 PF
// (dense) flagship rate. Opt-in screening precision (~1-2 significant
// digits; docs/NUMERICS.md); the engine default stays bf16.
//
// Operand semantics verified empirically by scripts/mx4_probe.hip:
// lane l carries 32 e2m1 nibbles (16 bytes, the LOW v4i of the v8i
// operand) of row/col l&31 with
//   hw_k(l, nib) = (l>>5)*16 + (nib & 15) + (nib >> 4)*32,
// and - UNLIKE the engine's fp8 model - the e8m0 scale applies PER
// SOURCE LANE (to the lane's own 32 nibbles; probe "scale-granularity"
// test). The fragment reads therefore load each lane's nibbles with
// ORIGINAL k CONTIGUOUS: nibble n of lane (row, kh) holds original
// k = kk*64 + kh*32 + n (one 16-B slot per (kk, kh): a single
// ds_read_b128). The induced hw_k permutation is identical on A and B,
// so products still pair equal original k, and the lane scale covers
// exactly original 32-block 4t + 2*kk + kh - the byte the kernel
// extracts from the packed scale word.
//
// Storage: A4/Bt4 row-major packed nibbles ([rows][K/2] bytes); scales
// packed-dword [K/128][rows] int32 (same layout as MX-fp8 round 2).
// Structure: the proven k128v3 shape - 256^2 tile, 8 waves (2Mx4N),
// K-tile = 128 (row = 64 B), double-buffered LDS (2 x 2 x 16 KiB =
// 64 KiB), one barrier per tile, scale words rotated one tile ahead.
#include "common.h"

extern unsigned g_st_m, g_st_n;  // supertile dims (gemm.hip)

typedef int v8i4 __attribute__((ext_vector_type(8)));
typedef int v2i4 __attribute__((ext_vector_type(2)));
typedef float f32x16q __attribute__((ext_vector_type(16)));

// build the v8i operand: payload in the LOW 4 regs (mx4_probe: hi=0)
__device__ __forceinline__ v8i4 v4_from16(const void* p) {
  typedef int v4i4 __attribute__((ext_vector_type(4)));
  v4i4 lo = *(const v4i4*)p;      // one ds_read_b128: nibbles 0-31
  v8i4 r = {};
  r[0] = lo[0]; r[1] = lo[1]; r[2] = lo[2]; r[3] = lo[3];
  return r;
}

#define M4BM 256
#define M4BN 256

__global__ void
__launch_bounds__(512)
gemm_bt_mx4_acc_k128(float* __restrict__ C,
                     const unsigned char* __restrict__ A,
                     const unsigned char* __restrict__ Asc,
                     const unsigned char* __restrict__ Bt,
                     const unsigned char* __restrict__ Bsc, int M, int N,
                     int K, int beta0, u32 stm_, u32 stn_) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  unsigned char* As = (unsigned char*)smem;  // [2][256*64]
  unsigned char* Bs = (unsigned char*)(smem + 2 * M4BM * 64);

  const u32 tid = threadIdx.x;
  const u32 lane = tid & 63, wave = tid >> 6;
  const u32 wm = wave >> 2, wn = wave & 3;

  const u32 ntm = (u32)(M / M4BM), ntn = (u32)(N / M4BN);
  u32 bid = xcd_swizzle(blockIdx.x, ntm * ntn);
  u32 tm, tn;
  if (stm_ && ntm % stm_ == 0 && ntn % stn_ == 0) {
    supertile_map(bid, ntm, ntn, stm_, stn_, &tm, &tn);
  } else {
    tm = bid / ntn;
    tn = bid % ntn;
  }
  const long m0 = (long)tm * M4BM, n0 = (long)tn * M4BN;
  const long KB2 = K / 2;  // packed row bytes

  // staging: tile = 256 rows x 64 B = 1024 x 16-B chunks; 512 threads
  // x 2 glds per operand (2 passes of 128 rows). 64-B rows = 4 slots;
  // swizzle slot ^ ((row>>2)&3) (the fp8 k64 pattern for 64-B rows).
  const u32 srow = tid >> 2;        // 0..127 (+p*128)
  const u32 sslot = tid & 3;
  const u32 src_slot = sslot ^ ((srow >> 2) & 3);

  auto stage = [&](int buf, int t) {
    const long k0b = (long)t * 64;  // K-tile of 128 = 64 packed bytes
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      u32 row = srow + half * 128;
      glds16(A + (m0 + row) * KB2 + k0b + src_slot * 16,
             As + buf * M4BM * 64 + row * 64u + sslot * 16u);
      glds16(Bt + (n0 + row) * KB2 + k0b + src_slot * 16,
             Bs + buf * M4BM * 64 + row * 64u + sslot * 16u);
    }
  };

  f32x16q acc[4][2] = {};
  const u32 arow = wm * 128 + (lane & 31);
  const u32 brow = wn * 64 + (lane & 31);
  const u32 kh = lane >> 5;

  auto load_scales = [&](int t, int* sa_w, int* sb_w) {
    // packed-dword scales [K/128][rows]: ONE coalesced load per word
#pragma unroll
    for (int m = 0; m < 4; ++m)
      sa_w[m] = ((const int*)Asc)[(long)t * M + m0 + arow + m * 32];
#pragma unroll
    for (int n = 0; n < 2; ++n)
      sb_w[n] = ((const int*)Bsc)[(long)t * N + n0 + brow + n * 32];
  };

  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  int sa_cur[4], sb_cur[2], sa_nxt[4], sb_nxt[2];
  load_scales(0, sa_cur, sb_cur);
  stage(0, 0);
  __syncthreads();
  int cur = 0;
  const int NT = K / 128;
  for (int t = 0; t < NT; ++t) {
    const bool pre = (t + 1 < NT);
    if (pre) {
      stage(cur ^ 1, t + 1);
      load_scales(t + 1, sa_nxt, sb_nxt);  // covered by the tile body
    }
    const unsigned char* a_base = As + cur * M4BM * 64;
    const unsigned char* b_base = Bs + cur * M4BM * 64;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      v8i4 af[4], bf[2];
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        u32 row = arow + m * 32;
        // slot = kk*2 + kh of the 64-B row (16 B = the lane's 32
        // contiguous original-k nibbles); slot bits XOR the staging
        // swizzle
        u32 byte = row * 64u + (((u32)kk * 2u + kh) ^ ((row >> 2) & 3))
                   * 16u;
        af[m] = v4_from16((const char*)a_base + byte);
      }
#pragma unroll
      for (int n = 0; n < 2; ++n) {
        u32 row = brow + n * 32;
        u32 byte = row * 64u + (((u32)kk * 2u + kh) ^ ((row >> 2) & 3))
                   * 16u;
        bf[n] = v4_from16((const char*)b_base + byte);
      }
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 2; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              af[m], bf[n], acc[m][n], 4, 4, 0,
              (sa_cur[m] >> (8 * (kk * 2 + (int)kh))) & 0xFF, 0,
              (sb_cur[n] >> (8 * (kk * 2 + (int)kh))) & 0xFF);
    }
    __syncthreads();
#pragma unroll
    for (int m = 0; m < 4; ++m) sa_cur[m] = sa_nxt[m];
#pragma unroll
    for (int n = 0; n < 2; ++n) sb_cur[n] = sb_nxt[n];
    cur ^= 1;
  }

  // epilogue: standard 32x32 C/D map, batched two-pass
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

extern "C" void launch_gemm_bt_mx4_acc(float* C, const void* A,
                                       const void* Asc, const void* Bt,
                                       const void* Bsc, int M, int N,
                                       int K, int beta0,
                                       hipStream_t stream) {
  dim3 grid((M / M4BM) * (N / M4BN));
  size_t lds = 4u * M4BM * 64;  // 64 KiB
  hipLaunchKernelGGL(gemm_bt_mx4_acc_k128, grid, dim3(512), lds, stream,
                     C, (const unsigned char*)A,
                     (const unsigned char*)Asc, (const unsigned char*)Bt,
                     (const unsigned char*)Bsc, M, N, K, beta0, g_st_m,
                     g_st_n);
}
