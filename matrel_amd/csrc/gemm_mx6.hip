// MX-fp6 (e2m3 + e8m0 block scales) GEMM for gfx950:
//   C(f32, MxN) += A6(MxK) @ Bt6(NxK)^T
// on v_mfma_scale_f32_32x32x64_f8f6f4 with cbsz=blgp=2. CDNA4 runs fp6
// at the FP4 MFMA rate - i.e. fp8-class accuracy (3 mantissa bits,
// ~6% per-element) at up to ~2x the fp8 GEMM rate. Opt-in
// (gemm_precision="mx6"); the engine default stays bf16.
//
// Operand semantics verified by scripts/mx6_probe.hip: payload = 24
// bytes/lane in the LOW 6 regs, LSB-first 6-bit fields, field k-map =
// the fp4 nibble map at field granularity, scale PER SOURCE LANE. As in
// the fp4 kernel, each lane's fields hold ORIGINAL k contiguous
// (k = kk*64 + kh*32 + field): the hw permutation cancels between A and
// B and the lane scale covers original 32-block 4t + 2*kk + kh.
//
// Storage (split-plane; see mxconvert.hip): each row of A6/Bt6 is
// [K/32 x 16B] (regs 0-3 of each block) then [K/32 x 8B] (regs 4-5);
// scales packed-dword [K/128][rows] int32. The split keeps both LDS
// read widths power-of-2:
//   region L: [row][4 x 16B slots], slot XOR-swizzled by (row>>2)&3
//     -> the mx4 kernel's proven conflict-free b128 pattern;
//   region H: [frag][256 rows][8B] mirroring the global H plane's
//     BLOCK-major order - staging reads each K-tile's H bytes as four
//     contiguous 2-KiB runs (full-128-B-line coalesced DMA; a
//     row-major H plane fragments into 32-B segments and cost ~25%
//     end to end - see profiles/README.md), and the b64 reads sit on
//     a conflict-free 2-dword lane stride with no swizzle at all.
// A 24-B-contiguous layout instead puts ALL reads at a 24-dword lane
// stride (2-way b128 + 4-way b64 conflicts) and measured only 1505 TF.
//
// Structure: the k128 shape - 256^2 tile, 8 waves, K-tile 128,
// double-buffered LDS (2 x 2 x (16 KiB + 8 KiB) = 96 KiB), one
// barrier per tile, scale rotation.
#include "common.h"
#include <cstdlib>

extern unsigned g_st_m, g_st_n;  // supertile dims (gemm.hip)

typedef int v8i6 __attribute__((ext_vector_type(8)));
typedef int v4i6 __attribute__((ext_vector_type(4)));
typedef int v2i6 __attribute__((ext_vector_type(2)));
typedef float f32x16h __attribute__((ext_vector_type(16)));

#define M6BM 256
#define M6BN 256
#define M6LROW 64   // region L row bytes (4 x 16B)
#define M6HFRAG 2048  // region H bytes per frag (256 rows x 8B)
#define M6LSZ (M6BM * M6LROW)
#define M6HSZ (4 * M6HFRAG)

template <int NOHI>
__device__ __forceinline__ v8i6 v6_read(const unsigned char* lbase,
                                        const unsigned char* hbase,
                                        u32 row, u32 frag) {
  u32 lslot = (frag ^ ((row >> 2) & 3u)) * 16u;
  v4i6 lo = *(const v4i6*)(lbase + row * M6LROW + lslot);  // ds_read_b128
  v8i6 r = {};
  r[0] = lo[0]; r[1] = lo[1]; r[2] = lo[2]; r[3] = lo[3];
  if (NOHI == 0) {  // NOHI>=1: perf probe only - result is WRONG (regs 4-5
                // zero); isolates the b64 stream's cost
    v2i6 hi = *(const v2i6*)(hbase + frag * M6HFRAG + row * 8u);  // ds_read_b64
    r[4] = hi[0]; r[5] = hi[1];
  }
  return r;
}

template <int NOHI>
__global__ void
__launch_bounds__(512)
gemm_bt_mx6_acc_k128(float* __restrict__ C,
                     const unsigned char* __restrict__ A,
                     const unsigned char* __restrict__ Asc,
                     const unsigned char* __restrict__ Bt,
                     const unsigned char* __restrict__ Bsc, int M, int N,
                     int K, int beta0, u32 stm_, u32 stn_) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  unsigned char* AsL = (unsigned char*)smem;            // [2][M6LSZ]
  unsigned char* BsL = (unsigned char*)(smem + 2 * M6LSZ);
  unsigned char* AsH = (unsigned char*)(smem + 4 * M6LSZ);  // [2][M6HSZ]
  unsigned char* BsH = (unsigned char*)(smem + 4 * M6LSZ + 2 * M6HSZ);

  const u32 tid = threadIdx.x;
  const u32 lane = tid & 63, wave = tid >> 6;
  const u32 wm = wave >> 2, wn = wave & 3;

  const u32 ntm = (u32)(M / M6BM), ntn = (u32)(N / M6BN);
  u32 bid = xcd_swizzle(blockIdx.x, ntm * ntn);
  u32 tm, tn;
  if (stm_ && ntm % stm_ == 0 && ntn % stn_ == 0) {
    supertile_map(bid, ntm, ntn, stm_, stn_, &tm, &tn);
  } else {
    tm = bid / ntn;
    tn = bid % ntn;
  }
  const long m0 = (long)tm * M6BM, n0 = (long)tn * M6BN;
  const long KL = (long)K / 2;       // L-plane bytes per row
  const long HA = (long)M * KL;      // A's H-plane base (block-major)
  const long HB = (long)N * KL;      // Bt's H-plane base

  // staging per operand: L = 256 rows x 4 16-B chunks (2 passes of
  // 512), H = 256 rows x 2 16-B chunks (1 pass). LDS destinations are
  // strictly LINEAR (glds16 writes wave-base + lane*16); the swizzles
  // permute the GLOBAL source chunk each lane fetches instead (XOR is
  // self-inverse, so the read-side slot formulas recover frag order).
  const u32 lrow = tid >> 2, lslot_st = tid & 3;
  const u32 lsrc = lslot_st ^ ((lrow >> 2) & 3u);
  const u32 hblk = tid >> 7, hi_ = tid & 127;  // chunk = rows 2i, 2i+1
  auto stage = [&](int buf, int t) {
    const long l0 = (long)t * 64;            // L bytes of this K-tile
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      u32 row = lrow + (u32)p * 128;
      glds16(A + (m0 + row) * KL + l0 + lsrc * 16,
             AsL + buf * M6LSZ + row * M6LROW + lslot_st * 16u);
      glds16(Bt + (n0 + row) * KL + l0 + lsrc * 16,
             BsL + buf * M6LSZ + row * M6LROW + lslot_st * 16u);
    }
    if (NOHI < 2) {  // NOHI=2 perf probe also drops the H staging DMA
      const long hb = (long)(4 * t + hblk);
      glds16(A + HA + (hb * M + m0 + hi_ * 2) * 8,
             AsH + buf * M6HSZ + tid * 16u);
      glds16(Bt + HB + (hb * N + n0 + hi_ * 2) * 8,
             BsH + buf * M6HSZ + tid * 16u);
    }
  };

  f32x16h acc[4][2] = {};
  const u32 arow = wm * 128 + (lane & 31);
  const u32 brow = wn * 64 + (lane & 31);
  const u32 kh = lane >> 5;

  auto load_scales = [&](int t, int* sa_w, int* sb_w) {
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
      load_scales(t + 1, sa_nxt, sb_nxt);
    }
    const unsigned char* al = AsL + cur * M6LSZ;
    const unsigned char* bl = BsL + cur * M6LSZ;
    const unsigned char* ah = AsH + cur * M6HSZ;
    const unsigned char* bh = BsH + cur * M6HSZ;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const u32 frag = (u32)kk * 2u + kh;
      v8i6 af[4], bf[2];
#pragma unroll
      for (int m = 0; m < 4; ++m)
        af[m] = v6_read<NOHI>(al, ah, arow + m * 32, frag);
#pragma unroll
      for (int n = 0; n < 2; ++n)
        bf[n] = v6_read<NOHI>(bl, bh, brow + n * 32, frag);
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 2; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              af[m], bf[n], acc[m][n], 2, 2, 0,
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

extern "C" void launch_gemm_bt_mx6_acc(float* C, const void* A,
                                       const void* Asc, const void* Bt,
                                       const void* Bsc, int M, int N,
                                       int K, int beta0,
                                       hipStream_t stream) {
  dim3 grid((M / M6BM) * (N / M6BN));
  size_t lds = 4u * M6LSZ + 4u * M6HSZ;  // 96 KiB
  const char* probe = getenv("MATREL_MX6_PROBE_NOHI");
  if (probe && probe[0] == '2')  // perf probe: WRONG results
    hipLaunchKernelGGL(gemm_bt_mx6_acc_k128<2>, grid, dim3(512), lds,
                       stream, C, (const unsigned char*)A,
                       (const unsigned char*)Asc, (const unsigned char*)Bt,
                       (const unsigned char*)Bsc, M, N, K, beta0, g_st_m,
                       g_st_n);
  else if (probe)  // perf probe: WRONG results
    hipLaunchKernelGGL(gemm_bt_mx6_acc_k128<1>, grid, dim3(512), lds,
                       stream, C, (const unsigned char*)A,
                       (const unsigned char*)Asc, (const unsigned char*)Bt,
                       (const unsigned char*)Bsc, M, N, K, beta0, g_st_m,
                       g_st_n);
  else
    hipLaunchKernelGGL(gemm_bt_mx6_acc_k128<0>, grid, dim3(512), lds,
                       stream, C, (const unsigned char*)A,
                       (const unsigned char*)Asc, (const unsigned char*)Bt,
                       (const unsigned char*)Bsc, M, N, K, beta0, g_st_m,
                       g_st_n);
}
