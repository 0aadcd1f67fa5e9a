// Dense MFMA GEMM kernels for gfx950 (MI355X).
//
// THE hot kernel of the engine (the reference's per-block
// Breeze/netlib dgemm becomes a hand-written CDNA4 MFMA GEMM).
//
// C(f32, M x N) += A(M x K) @ Bt(N x K)^T   - both operands row-major,
// B pre-transposed so both A- and B-tiles stage identically (contiguous
// K-rows; the B^T form is what the transpose-pushdown optimizer naturally
// produces, and summa.py transposes each broadcast panel once).
//
// Structure (cdna_hip_programming.md §5, the "step-3" ladder structure +
// T1/T2): 128x128 tile, BK=64, 4 waves (2x2), 16x16x32 bf16 MFMA, LDS
// double-buffered via 16-B global_load_lds with the XOR-16 source/read
// swizzle (rule 21: linear LDS dest, inverse-swizzled global source,
// swizzled ds_read), XCD-aware block swizzle.
//
// Shapes must satisfy M%128==0, N%128==0, K%64==0 (the Python wrapper pads
// ragged edge blocks).
#include "common.h"

#define BM 128
#define BN 128
#define BKH 64  // bf16 K-step
#define BKF 32  // f32  K-step
#define BM2 256
#define BN2 256
typedef float f32x16 __attribute__((ext_vector_type(16)));

// ---------------------------------------------------------------- bf16 ----
// LDS tiles: As[128][64] bf16 (16 KiB), Bs[128][64] bf16, x2 buffers = 64 KiB
// (2 blocks/CU fit in 160 KiB).

__global__ void
__launch_bounds__(256)
gemm_bt_bf16_acc_k(float* __restrict__ C, const bf16* __restrict__ A,
                   const bf16* __restrict__ Bt, int M, int N, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* As = (bf16*)smem;                       // [2][128*64]
  bf16* Bs = (bf16*)(smem + 2 * BM * BKH * 2);  // [2][128*64]

  const u32 tid = threadIdx.x;
  const u32 lane = tid & 63, wave = tid >> 6;
  const u32 wr = wave >> 1, wc = wave & 1;

  const u32 ntm = (u32)(M / BM), ntn = (u32)(N / BN);
  u32 bid = xcd_swizzle(blockIdx.x, ntm * ntn);
  const u32 tm = bid / ntn, tn = bid % ntn;
  const long m0 = (long)tm * BM, n0 = (long)tn * BN;

  // staging geometry: tile row = 64 cols * 2 B = 128 B = 8 16-B slots;
  // thread t stages slot (t%8) of row (t/8); source slot is XOR-swizzled.
  const u32 srow = tid >> 3, sslot = tid & 7;
  const u32 src_slot = sslot ^ (srow & 7);
  const int NT = K / BKH;

  auto stage = [&](int buf, int t) {
    const long k0 = (long)t * BKH;
    // 4 chunks of 32 rows each (256 threads x 16 B = 4 KiB per chunk)
    for (int it = 0; it < 4; ++it) {
      u32 row = srow + it * 32;
      const bf16* ga = A + (m0 + row) * (long)K + k0 + src_slot * 8;
      bf16* la = As + buf * BM * BKH + (row * 8u + sslot) * 8u;
      glds16(ga, la);
      const bf16* gb = Bt + (n0 + row) * (long)K + k0 + src_slot * 8;
      bf16* lb = Bs + buf * BM * BKH + (row * 8u + sslot) * 8u;
      glds16(gb, lb);
    }
  };

  f32x4 acc[4][4] = {};
  bf16x8 af[2][4], bf[2][4];

  const u32 arow_base = wr * 64 + (lane & 15);
  const u32 brow_base = wc * 64 + (lane & 15);
  const u32 kgrp = lane >> 4;  // 0..3: k sub-chunk of 8 within the 32-slice

  auto read_frags = [&](int buf) {
    const char* a_base = (const char*)(As + buf * BM * BKH);
    const char* b_base = (const char*)(Bs + buf * BM * BKH);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        u32 row = arow_base + m * 16;
        u32 byte = row * 128u + (u32)kk * 64u + kgrp * 16u;
        byte ^= (row & 7) << 4;
        af[kk][m] = *(const bf16x8*)(a_base + byte);
        u32 rowb = brow_base + m * 16;
        u32 byteb = rowb * 128u + (u32)kk * 64u + kgrp * 16u;
        byteb ^= (rowb & 7) << 4;
        bf[kk][m] = *(const bf16x8*)(b_base + byteb);
      }
  };

  stage(0, 0);
  __syncthreads();
  int cur = 0;
  for (int t = 0; t < NT; ++t) {
    if (t + 1 < NT) stage(cur ^ 1, t + 1);
    read_frags(cur);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[kk][m], bf[kk][n], acc[m][n], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();  // drains glds (vmcnt 0) + barrier
    cur ^= 1;
  }

  // epilogue: C(f32) += acc.  C/D map: col = lane&15, row = (lane>>4)*4+reg.
  // two-pass (batched loads, one wait) - see the 256p epilogue note.
  const u32 crow0 = wr * 64 + (lane >> 4) * 4;
  const u32 ccol0 = wc * 64 + (lane & 15);
#pragma unroll
  for (int m = 0; m < 4; ++m) {
    float cv[4][4];
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      long row = m0 + crow0 + m * 16;
      long col = n0 + ccol0 + n * 16;
      const float* cp = C + row * (long)N + col;
#pragma unroll
      for (int r = 0; r < 4; ++r) cv[n][r] = cp[(long)r * N];
    }
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      long row = m0 + crow0 + m * 16;
      long col = n0 + ccol0 + n * 16;
      float* cp = C + row * (long)N + col;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        cp[(long)r * N] = cv[n][r] + acc[m][n][r];
    }
  }
}

// ----------------------------------------------------------------- f32 ----
// Exact-f32 path (v_mfma_f32_16x16x4_f32: f32 in/acc at the 157 TF vector
// rate - cdna_hip_programming.md §3 "FP32-input MFMA"). Used for fp32
// matrices (solver outputs, OLS small multiplies) where bf16 rounding is
// not acceptable. Same tile structure; BK=32.

__global__ void
__launch_bounds__(256)
gemm_bt_f32_acc_k(float* __restrict__ C, const float* __restrict__ A,
                  const float* __restrict__ Bt, int M, int N, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* As = (float*)smem;                       // [2][128*32]
  float* Bs = (float*)(smem + 2 * BM * BKF * 4);  // [2][128*32]

  const u32 tid = threadIdx.x;
  const u32 lane = tid & 63, wave = tid >> 6;
  const u32 wr = wave >> 1, wc = wave & 1;

  const u32 ntm = (u32)(M / BM), ntn = (u32)(N / BN);
  u32 bid = xcd_swizzle(blockIdx.x, ntm * ntn);
  const u32 tm = bid / ntn, tn = bid % ntn;
  const long m0 = (long)tm * BM, n0 = (long)tn * BN;

  // tile row = 32 f32 = 128 B = 8 slots of 16 B (4 f32)
  const u32 srow = tid >> 3, sslot = tid & 7;
  const u32 src_slot = sslot ^ (srow & 7);
  const int NT = K / BKF;

  auto stage = [&](int buf, int t) {
    const long k0 = (long)t * BKF;
    for (int it = 0; it < 4; ++it) {
      u32 row = srow + it * 32;
      const float* ga = A + (m0 + row) * (long)K + k0 + src_slot * 4;
      float* la = As + buf * BM * BKF + (row * 8u + sslot) * 4u;
      glds16(ga, la);
      const float* gb = Bt + (n0 + row) * (long)K + k0 + src_slot * 4;
      float* lb = Bs + buf * BM * BKF + (row * 8u + sslot) * 4u;
      glds16(gb, lb);
    }
  };

  f32x4 acc[4][4] = {};
  const u32 arow_base = wr * 64 + (lane & 15);
  const u32 brow_base = wc * 64 + (lane & 15);
  const u32 kgrp = lane >> 4;  // k = k4*4 + kgrp

  stage(0, 0);
  __syncthreads();
  int cur = 0;
  for (int t = 0; t < NT; ++t) {
    if (t + 1 < NT) stage(cur ^ 1, t + 1);
    const char* a_base = (const char*)(As + cur * BM * BKF);
    const char* b_base = (const char*)(Bs + cur * BM * BKF);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int k4 = 0; k4 < BKF / 4; ++k4) {
      float a_s[4], b_s[4];
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        u32 row = arow_base + m * 16;
        u32 byte = row * 128u + (k4 * 4 + kgrp) * 4u;
        byte ^= (row & 7) << 4;
        a_s[m] = *(const float*)(a_base + byte);
        u32 rowb = brow_base + m * 16;
        u32 byteb = rowb * 128u + (k4 * 4 + kgrp) * 4u;
        byteb ^= (rowb & 7) << 4;
        b_s[m] = *(const float*)(b_base + byteb);
      }
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              a_s[m], b_s[n], acc[m][n], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
    cur ^= 1;
  }

  const u32 crow0 = wr * 64 + (lane >> 4) * 4;
  const u32 ccol0 = wc * 64 + (lane & 15);
#pragma unroll
  for (int m = 0; m < 4; ++m) {
    float cv[4][4];
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      long row = m0 + crow0 + m * 16;
      long col = n0 + ccol0 + n * 16;
      const float* cp = C + row * (long)N + col;
#pragma unroll
      for (int r = 0; r < 4; ++r) cv[n][r] = cp[(long)r * N];
    }
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      long row = m0 + crow0 + m * 16;
      long col = n0 + ccol0 + n * 16;
      float* cp = C + row * (long)N + col;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        cp[(long)r * N] = cv[n][r] + acc[m][n][r];
    }
  }
}

// ----------------------------------------------------- f32, 256^2 tile ---
// Exact-f32 on the 256^2/8-wave geometry (the 128^2 f32 kernel runs 2
// blocks/CU with per-K-tile barriers; the bigger tile halves the barrier
// rate per MFMA and doubles operand reuse). BK=32: LDS 2 buf x (A+B) x
// 256x32 f32 = 128 KiB; per-wave output 128x64 (8x4 16x16 frags) on
// v_mfma_f32_16x16x4_f32 (8 k4-steps per K-tile).

__global__ void
__launch_bounds__(512)
gemm_bt_f32_acc_256_k(float* __restrict__ C, const float* __restrict__ A,
                      const float* __restrict__ Bt, int M, int N, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* As = (float*)smem;                        // [2][256*32]
  float* Bs = (float*)(smem + 2 * BM2 * BKF * 4);  // [2][256*32]

  const u32 tid = threadIdx.x;
  const u32 lane = tid & 63, wave = tid >> 6;
  const u32 wm = wave >> 2, wn = wave & 3;  // 2 x 4 wave grid; 128x64 out

  const u32 ntm = (u32)(M / BM2), ntn = (u32)(N / BN2);
  u32 bid = xcd_swizzle(blockIdx.x, ntm * ntn);
  u32 tm, tn;
  if (ntm % 4 == 0 && ntn % 8 == 0) {
    supertile_map(bid, ntm, ntn, 4, 8, &tm, &tn);
  } else {
    tm = bid / ntn;
    tn = bid % ntn;
  }
  const long m0 = (long)tm * BM2, n0 = (long)tn * BN2;

  // staging: tile row = 32 f32 = 128 B = 8 slots; 256 rows x 8 slots =
  // 2048 chunks per operand; 512 threads x 4 passes x 2 operands.
  const u32 srow = tid >> 3, sslot = tid & 7;  // rows 0..63 per pass
  const u32 src_slot = sslot ^ (srow & 7);
  const int NT = K / BKF;

  auto stage = [&](int buf, int t) {
    const long k0 = (long)t * BKF;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      u32 row = srow + it * 64;
      glds16(A + (m0 + row) * (long)K + k0 + src_slot * 4,
             As + buf * BM2 * BKF + (row * 8u + sslot) * 4u);
      glds16(Bt + (n0 + row) * (long)K + k0 + src_slot * 4,
             Bs + buf * BM2 * BKF + (row * 8u + sslot) * 4u);
    }
  };

  f32x4 acc[8][4] = {};
  const u32 arow_base = wm * 128 + (lane & 15);
  const u32 brow_base = wn * 64 + (lane & 15);
  const u32 kgrp = lane >> 4;  // k = k4*4 + kgrp

  stage(0, 0);
  __syncthreads();
  int cur = 0;
  for (int t = 0; t < NT; ++t) {
    if (t + 1 < NT) stage(cur ^ 1, t + 1);
    const char* a_base = (const char*)(As + cur * BM2 * BKF);
    const char* b_base = (const char*)(Bs + cur * BM2 * BKF);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int k4 = 0; k4 < BKF / 4; ++k4) {
      float a_s[8], b_s[4];
#pragma unroll
      for (int m = 0; m < 8; ++m) {
        u32 row = arow_base + m * 16;
        u32 byte = row * 128u + (k4 * 4 + kgrp) * 4u;
        byte ^= (row & 7) << 4;
        a_s[m] = *(const float*)(a_base + byte);
      }
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        u32 row = brow_base + n * 16;
        u32 byte = row * 128u + (k4 * 4 + kgrp) * 4u;
        byte ^= (row & 7) << 4;
        b_s[n] = *(const float*)(b_base + byte);
      }
#pragma unroll
      for (int m = 0; m < 8; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              a_s[m], b_s[n], acc[m][n], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
    cur ^= 1;
  }

  const u32 crow0 = wm * 128 + (lane >> 4) * 4;
  const u32 ccol0 = wn * 64 + (lane & 15);
#pragma unroll
  for (int m = 0; m < 8; ++m) {
    float cv[4][4];
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      long row = m0 + crow0 + m * 16;
      long col = n0 + ccol0 + n * 16;
      const float* cp = C + row * (long)N + col;
#pragma unroll
      for (int r = 0; r < 4; ++r) cv[n][r] = cp[(long)r * N];
    }
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      long row = m0 + crow0 + m * 16;
      long col = n0 + ccol0 + n * 16;
      float* cp = C + row * (long)N + col;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        cp[(long)r * N] = cv[n][r] + acc[m][n][r];
    }
  }
}

// ------------------------------------ f32, 256^2 tile, 32x32x2 MFMA ------
// The 16x16x4 f32 MFMA is only ~8 pipe-cycles per instruction - at the
// 4-cycle wave issue cadence plus fragment reads the f32 kernels above
// are ISSUE-bound (~73% of the 157 TF f32 rate). v_mfma_f32_32x32x2_f32
// does 2x the flops per instruction (4096), halving the MFMA instruction
// count per tile: per-wave 128x64 output = 4x2 32x32 fragments,
// 16 k2-steps per BK=32 tile, 6 b32 fragment reads per step.
// C/D map (32x32): col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5).

__global__ void
__launch_bounds__(512)
gemm_bt_f32_acc_256x32_k(float* __restrict__ C, const float* __restrict__ A,
                         const float* __restrict__ Bt, int M, int N,
                         int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* As = (float*)smem;                        // [2][256*32]
  float* Bs = (float*)(smem + 2 * BM2 * BKF * 4);  // [2][256*32]

  const u32 tid = threadIdx.x;
  const u32 lane = tid & 63, wave = tid >> 6;
  const u32 wm = wave >> 2, wn = wave & 3;  // 2 x 4 grid; 128x64 per wave

  const u32 ntm = (u32)(M / BM2), ntn = (u32)(N / BN2);
  u32 bid = xcd_swizzle(blockIdx.x, ntm * ntn);
  u32 tm, tn;
  if (ntm % 4 == 0 && ntn % 8 == 0) {
    supertile_map(bid, ntm, ntn, 4, 8, &tm, &tn);
  } else {
    tm = bid / ntn;
    tn = bid % ntn;
  }
  const long m0 = (long)tm * BM2, n0 = (long)tn * BN2;

  const u32 srow = tid >> 3, sslot = tid & 7;
  const u32 src_slot = sslot ^ (srow & 7);
  const int NT = K / BKF;

  auto stage = [&](int buf, int t) {
    const long k0 = (long)t * BKF;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      u32 row = srow + it * 64;
      glds16(A + (m0 + row) * (long)K + k0 + src_slot * 4,
             As + buf * BM2 * BKF + (row * 8u + sslot) * 4u);
      glds16(Bt + (n0 + row) * (long)K + k0 + src_slot * 4,
             Bs + buf * BM2 * BKF + (row * 8u + sslot) * 4u);
    }
  };

  f32x16 acc[4][2] = {};  // [m-frag 32][n-frag 32]
  const u32 arow_base = wm * 128 + (lane & 31);
  const u32 brow_base = wn * 64 + (lane & 31);
  const u32 kh = lane >> 5;  // k = k2*2 + kh

  stage(0, 0);
  __syncthreads();
  int cur = 0;
  for (int t = 0; t < NT; ++t) {
    if (t + 1 < NT) stage(cur ^ 1, t + 1);
    const char* a_base = (const char*)(As + cur * BM2 * BKF);
    const char* b_base = (const char*)(Bs + cur * BM2 * BKF);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int k2 = 0; k2 < BKF / 2; ++k2) {
      float a_s[4], b_s[2];
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        u32 row = arow_base + m * 32;
        u32 byte = row * 128u + (k2 * 2 + kh) * 4u;
        byte ^= (row & 7) << 4;
        a_s[m] = *(const float*)(a_base + byte);
      }
#pragma unroll
      for (int n = 0; n < 2; ++n) {
        u32 row = brow_base + n * 32;
        u32 byte = row * 128u + (k2 * 2 + kh) * 4u;
        byte ^= (row & 7) << 4;
        b_s[n] = *(const float*)(b_base + byte);
      }
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 2; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_32x32x2f32(
              a_s[m], b_s[n], acc[m][n], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
    cur ^= 1;
  }

  // epilogue: 32x32 C/D map, batched two-pass
  const long ccol = n0 + wn * 64 + (lane & 31);
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      float cv[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        long row = m0 + wm * 128 + m * 32 + (r & 3) + 8 * (r >> 2) +
                   4 * (lane >> 5);
        cv[r] = C[row * (long)N + ccol + n * 32];
      }
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        long row = m0 + wm * 128 + m * 32 + (r & 3) + 8 * (r >> 2) +
                   4 * (lane >> 5);
        C[row * (long)N + ccol + n * 32] = cv[r] + acc[m][n][r];
      }
    }
}

// --------------------------------------------------- bf16, 256^2 tile ----
// 8 waves (2M x 4N), per-wave 128x64 output (8x4 16x16 frags), BK=64,
// 2 LDS buffers x (A 256x64 + B 256x64) = 128 KiB -> 1 block/CU.
// The cdna_hip_programming.md §5 table row: "glds, 2 LDS buffers, BK=64,
// vmcnt(0) + plain __syncthreads()" - the top BK=64 tier for this tile.

__global__ void
__launch_bounds__(512)
gemm_bt_bf16_acc_256_k(float* __restrict__ C, const bf16* __restrict__ A,
                       const bf16* __restrict__ Bt, int M, int N, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* As = (bf16*)smem;                        // [2][256*64]
  bf16* Bs = (bf16*)(smem + 2 * BM2 * BKH * 2);  // [2][256*64]

  const u32 tid = threadIdx.x;
  const u32 lane = tid & 63, wave = tid >> 6;
  const u32 wm = wave >> 2, wn = wave & 3;  // 2 x 4 wave grid

  const u32 ntm = (u32)(M / BM2), ntn = (u32)(N / BN2);
  u32 bid = xcd_swizzle(blockIdx.x, ntm * ntn);
  u32 tm, tn;
  if (ntm % 4 == 0 && ntn % 8 == 0) {
    supertile_map(bid, ntm, ntn, 4, 8, &tm, &tn);
  } else {
    tm = bid / ntn;
    tn = bid % ntn;
  }
  const long m0 = (long)tm * BM2, n0 = (long)tn * BN2;

  // staging: tile = 256 rows x 64 cols x 2 B; row = 8 x 16-B slots.
  // 512 threads x 16 B = 8 KiB per pass; 4 passes per operand tile.
  const u32 srow = tid >> 3, sslot = tid & 7;  // rows 0..63 per pass
  const u32 src_slot = sslot ^ (srow & 7);
  const int NT = K / BKH;

  auto stage = [&](int buf, int t) {
    const long k0 = (long)t * BKH;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      u32 row = srow + it * 64;
      const bf16* ga = A + (m0 + row) * (long)K + k0 + src_slot * 8;
      glds16(ga, As + buf * BM2 * BKH + (row * 8u + sslot) * 8u);
      const bf16* gb = Bt + (n0 + row) * (long)K + k0 + src_slot * 8;
      glds16(gb, Bs + buf * BM2 * BKH + (row * 8u + sslot) * 8u);
    }
  };

  f32x4 acc[8][4] = {};
  const u32 arow_base = wm * 128 + (lane & 15);
  const u32 brow_base = wn * 64 + (lane & 15);
  const u32 kgrp = lane >> 4;

  stage(0, 0);
  __syncthreads();
  int cur = 0;
  for (int t = 0; t < NT; ++t) {
    if (t + 1 < NT) stage(cur ^ 1, t + 1);
    const char* a_base = (const char*)(As + cur * BM2 * BKH);
    const char* b_base = (const char*)(Bs + cur * BM2 * BKH);
    bf16x8 af[8], bf[4];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int m = 0; m < 8; ++m) {
        u32 row = arow_base + m * 16;
        u32 byte = row * 128u + (u32)kk * 64u + kgrp * 16u;
        byte ^= (row & 7) << 4;
        af[m] = *(const bf16x8*)(a_base + byte);
      }
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        u32 rowb = brow_base + n * 16;
        u32 byteb = rowb * 128u + (u32)kk * 64u + kgrp * 16u;
        byteb ^= (rowb & 7) << 4;
        bf[n] = *(const bf16x8*)(b_base + byteb);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int m = 0; m < 8; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[m], bf[n], acc[m][n], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
    cur ^= 1;
  }

  const u32 crow0 = wm * 128 + (lane >> 4) * 4;
  const u32 ccol0 = wn * 64 + (lane & 15);
#pragma unroll
  for (int m = 0; m < 8; ++m) {
    float cv[4][4];
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      long row = m0 + crow0 + m * 16;
      long col = n0 + ccol0 + n * 16;
      const float* cp = C + row * (long)N + col;
#pragma unroll
      for (int r = 0; r < 4; ++r) cv[n][r] = cp[(long)r * N];
    }
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      long row = m0 + crow0 + m * 16;
      long col = n0 + ccol0 + n * 16;
      float* cp = C + row * (long)N + col;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        cp[(long)r * N] = cv[n][r] + acc[m][n][r];
    }
  }
}

// ------------------------------------- bf16, 256^2 tile, 4-phase pipe ----
// The counted-vmcnt phase pipeline (cdna_hip_programming.md §5 "The 256^2
// 8-phase template", T3+T4+T5): per K-tile, the 8 waves TOGETHER compute
// one 128x128 C-quadrant per phase over the full K=64, so every wave
// consumes the tile's four 16-KiB half-tiles (A0,B0,B1,A1) in the SAME
// order. Each phase stages ONE half-tile of tile t+1 (2 glds/thread) and
// ends with s_waitcnt vmcnt(4) + raw s_barrier - the glds queue is never
// drained to 0 inside the loop (the last two staged halves stay in
// flight across every barrier). Quadrant snake order (0,0)->(0,1)->
// (1,1)->(1,0) reuses one operand's fragments at each transition.

// DEEPRING: stage ALL FOUR halves of tile t+1 during phases 1-2 of tile t
// (legal: buffer nxt's last reader is tile t-1, whose final read of any
// half region is >=1 barrier before t.p1). Every vmcnt then proves a load
// issued >=4 phases earlier, so the waits never actually park the wave -
// counted ring per the 8-phase template's "vmcnt never 0 in the main
// loop" rule. Steady-state waits: p1-end vmcnt(6) proves B1(cur),
// p2-end vmcnt(8) proves A1(cur), p4-end vmcnt(4) proves A0',B0'.
// SCHED: split each phase's reads+MFMAs by kk half and pin the shape with
// sched_group_barrier so the kk=0 MFMA cluster starts after only its own
// 6 LDS reads land (the compiler's conservative lgkmcnt(0) then covers
// exactly those 6); the kk=1 reads' latency hides under the kk=0 MFMAs.
template <bool MIDBAR, bool STATICPRIO = false, bool BETA0 = false,
          bool BF16OUT = false, bool DEEPRING = false, bool SCHED = false>
__global__ void
__launch_bounds__(512)
gemm_bt_bf16_acc_256p_k(float* __restrict__ C, const bf16* __restrict__ A,
                        const bf16* __restrict__ Bt, int M, int N, int K,
                        u32 stm_, u32 stn_, bf16* __restrict__ Co = nullptr) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // layout: buf b (2) x half h (0=A0 1=A1 2=B0 3=B1) x 16 KiB; all LDS
  // addressing is 32-bit offsets from one base (keeps VGPR pressure down).
#define HOFF(b, h) (((b) * 4u + (h)) * (128u * BKH * 2u))

  const u32 tid = threadIdx.x;
  const u32 lane = tid & 63, wave = tid >> 6;
  // within a quadrant: wave w -> 32-row x 64-col slice
  const u32 qr = (wave & 3) * 32;   // row offset in the 128x128 quadrant
  const u32 qc = (wave >> 2) * 64;  // col offset

  const u32 ntm = (u32)(M / BM2), ntn = (u32)(N / BN2);
  u32 bid = xcd_swizzle(blockIdx.x, ntm * ntn);
  u32 tm, tn;
  if (stm_ && ntm % stm_ == 0 && ntn % stn_ == 0) {
    supertile_map(bid, ntm, ntn, stm_, stn_, &tm, &tn);
  } else {
    tm = bid / ntn;
    tn = bid % ntn;
  }
  const long m0 = (long)tm * BM2, n0 = (long)tn * BN2;

  // staging: one half-tile = 128 rows x 64 cols = 1024 x 16 B; 512 threads
  // x 2 glds. Same XOR-16 source swizzle as the other kernels.
  const u32 srow = tid >> 3, sslot = tid & 7;  // rows 0..63
  const u32 src_slot = sslot ^ (srow & 7);
  const int NT = K / BKH;

  // stage half h (0=A0,1=A1,2=B0,3=B1) of K-tile t into buffer buf.
  // h is a literal at every call site, so src/dst fold at compile time.
  auto stage_half = [&](u32 bufoff, int h, int t) {
    const long k0 = (long)t * BKH;
    const bf16* src = (h < 2)
        ? A + (m0 + (h & 1) * 128) * (long)K
        : Bt + (n0 + (h & 1) * 128) * (long)K;
    bf16* dst = (bf16*)(smem + bufoff + (u32)h * (128u * BKH * 2u));
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      u32 row = srow + it * 64;
      glds16(src + row * (long)K + k0 + src_slot * 8,
             dst + (row * 8u + sslot) * 8u);
    }
  };

  f32x4 acc[4][2][4] = {};  // [quadrant][m:2][n:4]
  bf16x8 af[2][2], bfr[2][4];  // [kk][m] / [kk][n]

  const u32 arow_base = qr + (lane & 15);
  const u32 brow_base = qc + (lane & 15);
  const u32 kgrp = lane >> 4;

  auto read_a = [&](u32 off) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int m = 0; m < 2; ++m) {
        u32 row = arow_base + m * 16;
        u32 byte = row * 128u + (u32)kk * 64u + kgrp * 16u;
        byte ^= (row & 7) << 4;
        af[kk][m] = *(const bf16x8*)(smem + off + byte);
      }
  };
  auto read_b = [&](u32 off) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        u32 row = brow_base + n * 16;
        u32 byte = row * 128u + (u32)kk * 64u + kgrp * 16u;
        byte ^= (row & 7) << 4;
        bfr[kk][n] = *(const bf16x8*)(smem + off + byte);
      }
  };
  // SCHED: per-kk fragment reads, so each phase can be hard-split with
  // sched_barrier(0) fences into {kk0 reads + stage | kk0 MFMA cluster |
  // kk1 reads | kk1 MFMA cluster}: the compiler's conservative lgkmcnt(0)
  // before the kk0 cluster then covers only the 2-6 kk0 reads (instead of
  // all 12), and the kk1 reads' LDS latency hides under the kk0 MFMAs.
  auto read_a_kk = [&](u32 off, int kk) {
#pragma unroll
    for (int m = 0; m < 2; ++m) {
      u32 row = arow_base + m * 16;
      u32 byte = row * 128u + (u32)kk * 64u + kgrp * 16u;
      byte ^= (row & 7) << 4;
      af[kk][m] = *(const bf16x8*)(smem + off + byte);
    }
  };
  auto read_b_kk = [&](u32 off, int kk) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      u32 row = brow_base + n * 16;
      u32 byte = row * 128u + (u32)kk * 64u + kgrp * 16u;
      byte ^= (row & 7) << 4;
      bfr[kk][n] = *(const bf16x8*)(smem + off + byte);
    }
  };

#define MFMA_KK(q, kk)                                                     \
  _Pragma("unroll") for (int m = 0; m < 2; ++m)                            \
      _Pragma("unroll") for (int n = 0; n < 4; ++n)                        \
          acc[q][m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(          \
              af[kk][m], bfr[kk][n], acc[q][m][n], 0, 0, 0);

#define MFMA_QUAD(q)                                                       \
  if (!STATICPRIO) __builtin_amdgcn_s_setprio(1);                          \
  _Pragma("unroll") for (int kk = 0; kk < 2; ++kk)                         \
      _Pragma("unroll") for (int m = 0; m < 2; ++m)                        \
          _Pragma("unroll") for (int n = 0; n < 4; ++n)                    \
              acc[q][m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(      \
                  af[kk][m], bfr[kk][n], acc[q][m][n], 0, 0, 0);           \
  if (!STATICPRIO) __builtin_amdgcn_s_setprio(0);

  // T5 static form: one priority raise for the second-dispatched half
  // (the arbitration loser on every segment), no per-cluster flips.
  // Condition must be provably wave-uniform (readfirstlane) or the
  // setprio applies to every wave (cdna_hip_programming.md T5 recipe).
  if (STATICPRIO &&
      __builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  // prologue: stage tile 0 completely in consumption-priority order
  stage_half(0u, 0, 0);  // A0
  stage_half(0u, 2, 0);  // B0
  stage_half(0u, 3, 0);  // B1
  stage_half(0u, 1, 0);  // A1
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");  // A0,B0 landed
  __builtin_amdgcn_s_barrier();

  const u32 BUFSZ = 4u * 128u * BKH * 2u;
  for (int t = 0; t < NT; ++t) {
    const u32 cur = (u32)(t & 1) * BUFSZ, nxt = cur ^ BUFSZ;
    const bool pre = (t + 1 < NT);
    // ---- phase 1: Q(0,0) - reads A0,B0; stages A0 of t+1 (+B0' if DEEP)
    if constexpr (SCHED) {
      read_a_kk(cur + HOFF(0, 0), 0);
      read_b_kk(cur + HOFF(0, 2), 0);
      if (pre) stage_half(nxt, 0, t + 1);
      __builtin_amdgcn_sched_barrier(0);
      MFMA_KK(0, 0)
      __builtin_amdgcn_sched_barrier(0);
      read_a_kk(cur + HOFF(0, 0), 1);
      read_b_kk(cur + HOFF(0, 2), 1);
      __builtin_amdgcn_sched_barrier(0);
      MFMA_KK(0, 1)
    } else {
      read_a(cur + HOFF(0, 0));
      read_b(cur + HOFF(0, 2));
      if (pre) stage_half(nxt, 0, t + 1);
      if (DEEPRING && pre) stage_half(nxt, 2, t + 1);
      if (MIDBAR) __builtin_amdgcn_s_barrier();
      MFMA_QUAD(0)
    }
    // drain B1(t); leave {A1(t), A0'} (or {A1} when not staging)
    if (DEEPRING) {
      if (pre) asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
      else     asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    } else {
      if (pre) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      else     asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    // ---- phase 2: Q(0,1) - reads B1 (A frags reused); stages B0'
    // (DEEP: B1'+A1' - tile t+1 fully issued by end of this phase)
    if constexpr (SCHED) {
      read_b_kk(cur + HOFF(0, 3), 0);
      if (pre) stage_half(nxt, 2, t + 1);
      __builtin_amdgcn_sched_barrier(0);
      MFMA_KK(1, 0)
      __builtin_amdgcn_sched_barrier(0);
      read_b_kk(cur + HOFF(0, 3), 1);
      __builtin_amdgcn_sched_barrier(0);
      MFMA_KK(1, 1)
    } else {
      read_b(cur + HOFF(0, 3));
      if (DEEPRING) {
        if (pre) { stage_half(nxt, 3, t + 1); stage_half(nxt, 1, t + 1); }
      } else {
        if (pre) stage_half(nxt, 2, t + 1);
      }
      if (MIDBAR) __builtin_amdgcn_s_barrier();
      MFMA_QUAD(1)
    }
    // drain A1(t)
    if (DEEPRING) {
      if (pre) asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      else     asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    } else {
      if (pre) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      else     asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    // ---- phase 3: Q(1,1) - reads A1 (B frags reused); stages B1'
    if constexpr (SCHED) {
      read_a_kk(cur + HOFF(0, 1), 0);
      if (pre) stage_half(nxt, 3, t + 1);
      __builtin_amdgcn_sched_barrier(0);
      MFMA_KK(2, 0)
      __builtin_amdgcn_sched_barrier(0);
      read_a_kk(cur + HOFF(0, 1), 1);
      __builtin_amdgcn_sched_barrier(0);
      MFMA_KK(2, 1)
    } else {
      read_a(cur + HOFF(0, 1));
      if (!DEEPRING && pre) stage_half(nxt, 3, t + 1);
      if (MIDBAR) __builtin_amdgcn_s_barrier();
      MFMA_QUAD(2)
    }
    // no barrier here: phase 4 reads B0(cur) - landed and barriered at
    // the tile boundary - and stages into the OTHER buffer.
    if (MIDBAR) __builtin_amdgcn_s_barrier();
    // ---- phase 4: Q(1,0) - re-reads B0 (landed long ago); stages A1'
    if constexpr (SCHED) {
      read_b_kk(cur + HOFF(0, 2), 0);
      if (pre) stage_half(nxt, 1, t + 1);
      __builtin_amdgcn_sched_barrier(0);
      MFMA_KK(3, 0)
      __builtin_amdgcn_sched_barrier(0);
      read_b_kk(cur + HOFF(0, 2), 1);
      __builtin_amdgcn_sched_barrier(0);
      MFMA_KK(3, 1)
    } else {
      read_b(cur + HOFF(0, 2));
      if (!DEEPRING && pre) stage_half(nxt, 1, t + 1);
      if (MIDBAR) __builtin_amdgcn_s_barrier();
      MFMA_QUAD(3)
    }
    // end of tile: drain up to {B1', A1'} for next P1
    if (pre) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  // epilogue: C(f32) += acc; quadrant (mh,nh) at rows mh*128, cols nh*128.
  // Two-pass per quadrant (gather 32 C values, then add+store) so the
  // loads batch under ONE vmcnt instead of 128 serialized
  // load->vmcnt(0)->add->store chains.
  const u32 crow0 = qr + (lane >> 4) * 4;
  const u32 ccol0 = qc + (lane & 15);
  const int QMH[4] = {0, 0, 1, 1};
  const int QNH[4] = {0, 1, 1, 0};
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    float cv[2][4][4];
    if (!BETA0) {
#pragma unroll
      for (int m = 0; m < 2; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n) {
          long row = m0 + QMH[q] * 128 + crow0 + m * 16;
          long col = n0 + QNH[q] * 128 + ccol0 + n * 16;
          const float* cp = C + row * (long)N + col;
#pragma unroll
          for (int r = 0; r < 4; ++r) cv[m][n][r] = cp[(long)r * N];
        }
    }
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        long row = m0 + QMH[q] * 128 + crow0 + m * 16;
        long col = n0 + QNH[q] * 128 + ccol0 + n * 16;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = (BETA0 ? 0.0f : cv[m][n][r]) + acc[q][m][n][r];
          if (BF16OUT)
            Co[(row + r) * (long)N + col] = (bf16)v;
          else
            C[(row + r) * (long)N + col] = v;
        }
      }
  }
}
#undef MFMA_QUAD
#undef MFMA_KK
#undef HOFF

// ----------------- bf16, 256^2 tile, 2-barrier pipelined ring (v14) ------
// Round-2 restructure of the phased kernel, targeting the 25.2% parked
// fraction rocprof showed on variant 6 (profiles/README.md bf16v6_16k_pmc):
// the 4-phase kernel pays 4 barriers per K-tile and starts EVERY phase
// with a 4-12 ds_read burst + lgkmcnt drain during which the SIMD's MFMA
// pipe idles. Two structural changes, same register plan (48 fragment
// VGPRs, B0/B1 re-read in the second half exactly like variant 6):
//
//  1. TWO barriers per tile instead of four. All 8 staging glds of tile
//     t+1 issue during the first half of tile t; the mid-tile barrier M
//     (own-wave vmcnt(0) first) proves the whole of tile t+1 landed, so
//     the second half may read ANY of it; the end barrier B is a bare
//     WAR fence (nobody restages buffer `cur` until every wave passed B).
//  2. Every 8-MFMA cluster runs on fragments read ONE cluster earlier
//     (software pipelining at cluster granularity, sched_barrier(0)
//     fenced): cluster j's lgkmcnt is satisfied by the time cluster j-1's
//     512+ cycles of MFMA pipe time have drained, so no cluster head ever
//     parks on LDS. PF0 additionally prefetches the next TILE's head
//     fragments (A0',B0' kk0) between M and B - legal exactly because M
//     proved the nxt buffer - so even the tile boundary starts on MFMAs.
//
// Cluster stream per tile (Q0=C(0,0) Q1=C(0,1) Q2=C(1,1) Q3=C(1,0);
// af[kk][2] A-frags, bfr[kk][4] B-frags, rotating in place):
//   C0:Q0kk0(a0,b0) | stage A0'B0' | R:a1,b1 | C1:Q0kk1 | stage B1'A1' |
//   R:b2->bfr0 | C2:Q1kk0 | R:b3->bfr1 | C3:Q1kk1 | R:a2->af0 |
//   vmcnt(0) BARRIER-M | C4:Q2kk0 | R:a3->af1 | C5:Q2kk1 | R:b0->bfr0 |
//   C6:Q3kk0 | R:b1->bfr1 | C7:Q3kk1 | PF0: R:a0',b0' | BARRIER-B

// ONEBAR (variant 16): drop the mid-tile barrier entirely - the measured
// failure of the 2-barrier form (947 vs 1199 TF, gpurun r2 call 1) was
// the mid-tile vmcnt(0): staging glds issued ~1300 cycles earlier parked
// every wave under loaded-HBM latency. With ONE barrier per tile the
// proof moves to the tile-end barrier (~3000+ cycles of slack for the
// same glds) and the cross-tile prefetch (PF0) is dropped because the
// nxt buffer is only proven at B; the tile head pays one ~100-cycle LDS
// drain instead.
template <bool BETA0 = false, bool BF16OUT = false, bool PF0 = true,
          bool ONEBAR = false>
__global__ void
__launch_bounds__(512)
gemm_bt_bf16_acc_256r_k(float* __restrict__ C, const bf16* __restrict__ A,
                        const bf16* __restrict__ Bt, int M, int N, int K,
                        u32 stm_, u32 stn_, bf16* __restrict__ Co = nullptr) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
#define HOFFR(b, h) (((b) * 4u + (h)) * (128u * BKH * 2u))

  const u32 tid = threadIdx.x;
  const u32 lane = tid & 63, wave = tid >> 6;
  const u32 qr = (wave & 3) * 32;
  const u32 qc = (wave >> 2) * 64;

  const u32 ntm = (u32)(M / BM2), ntn = (u32)(N / BN2);
  u32 bid = xcd_swizzle(blockIdx.x, ntm * ntn);
  u32 tm, tn;
  if (stm_ && ntm % stm_ == 0 && ntn % stn_ == 0) {
    supertile_map(bid, ntm, ntn, stm_, stn_, &tm, &tn);
  } else {
    tm = bid / ntn;
    tn = bid % ntn;
  }
  const long m0 = (long)tm * BM2, n0 = (long)tn * BN2;

  const u32 srow = tid >> 3, sslot = tid & 7;
  const u32 src_slot = sslot ^ (srow & 7);
  const int NT = K / BKH;

  auto stage_half = [&](u32 bufoff, int h, int t) {
    const long k0 = (long)t * BKH;
    const bf16* src = (h < 2)
        ? A + (m0 + (h & 1) * 128) * (long)K
        : Bt + (n0 + (h & 1) * 128) * (long)K;
    bf16* dst = (bf16*)(smem + bufoff + (u32)h * (128u * BKH * 2u));
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      u32 row = srow + it * 64;
      glds16(src + row * (long)K + k0 + src_slot * 8,
             dst + (row * 8u + sslot) * 8u);
    }
  };

  f32x4 acc[4][2][4] = {};
  bf16x8 af[2][2], bfr[2][4];

  const u32 arow_base = qr + (lane & 15);
  const u32 brow_base = qc + (lane & 15);
  const u32 kgrp = lane >> 4;

  // slot: which half of the rotating register set to write (0/1);
  // kk: which 32-wide k sub-slice of the LDS half to read.
  auto read_a_s = [&](u32 off, int slot, int kk) {
#pragma unroll
    for (int m = 0; m < 2; ++m) {
      u32 row = arow_base + m * 16;
      u32 byte = row * 128u + (u32)kk * 64u + kgrp * 16u;
      byte ^= (row & 7) << 4;
      af[slot][m] = *(const bf16x8*)(smem + off + byte);
    }
  };
  auto read_b_s = [&](u32 off, int slot, int kk) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      u32 row = brow_base + n * 16;
      u32 byte = row * 128u + (u32)kk * 64u + kgrp * 16u;
      byte ^= (row & 7) << 4;
      bfr[slot][n] = *(const bf16x8*)(smem + off + byte);
    }
  };

#define MFMA_RS(q, s)                                                      \
  _Pragma("unroll") for (int m = 0; m < 2; ++m)                            \
      _Pragma("unroll") for (int n = 0; n < 4; ++n)                        \
          acc[q][m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(          \
              af[s][m], bfr[s][n], acc[q][m][n], 0, 0, 0);
#define SFENCE() __builtin_amdgcn_sched_barrier(0)

  // static priority for the second-dispatched half (T5 static form)
  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  // prologue: stage tile 0 fully, prove it, preload the head fragments
  stage_half(0u, 0, 0);
  stage_half(0u, 2, 0);
  stage_half(0u, 3, 0);
  stage_half(0u, 1, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  read_a_s(HOFFR(0, 0), 0, 0);  // a0
  read_b_s(HOFFR(0, 2), 0, 0);  // b0

  const u32 BUFSZ = 4u * 128u * BKH * 2u;
  for (int t = 0; t < NT; ++t) {
    const u32 cur = (u32)(t & 1) * BUFSZ, nxt = cur ^ BUFSZ;
    const bool pre = (t + 1 < NT);
    if (!PF0 && t > 0) {
      read_a_s(cur + HOFFR(0, 0), 0, 0);
      read_b_s(cur + HOFFR(0, 2), 0, 0);
    }
    SFENCE();
    MFMA_RS(0, 0)  // C0
    SFENCE();
    if (pre) { stage_half(nxt, 0, t + 1); stage_half(nxt, 2, t + 1); }
    read_a_s(cur + HOFFR(0, 0), 1, 1);  // a1
    read_b_s(cur + HOFFR(0, 2), 1, 1);  // b1
    SFENCE();
    MFMA_RS(0, 1)  // C1
    SFENCE();
    if (pre) { stage_half(nxt, 3, t + 1); stage_half(nxt, 1, t + 1); }
    read_b_s(cur + HOFFR(0, 3), 0, 0);  // b2 -> bfr[0]
    SFENCE();
    MFMA_RS(1, 0)  // C2 (a0 x B1kk0)
    SFENCE();
    read_b_s(cur + HOFFR(0, 3), 1, 1);  // b3 -> bfr[1]
    SFENCE();
    MFMA_RS(1, 1)  // C3
    SFENCE();
    read_a_s(cur + HOFFR(0, 1), 0, 0);  // a2 -> af[0] (A1kk0, own tile)
    if constexpr (!ONEBAR) {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // staging landed
      __builtin_amdgcn_s_barrier();                     // M
    }
    MFMA_RS(2, 0)  // C4 (A1kk0 x B1kk0)
    SFENCE();
    read_a_s(cur + HOFFR(0, 1), 1, 1);  // a3 -> af[1]
    SFENCE();
    MFMA_RS(2, 1)  // C5
    SFENCE();
    read_b_s(cur + HOFFR(0, 2), 0, 0);  // b0 re-read -> bfr[0]
    SFENCE();
    MFMA_RS(3, 0)  // C6 (A1kk0 x B0kk0)
    SFENCE();
    read_b_s(cur + HOFFR(0, 2), 1, 1);  // b1 re-read -> bfr[1]
    SFENCE();
    MFMA_RS(3, 1)  // C7
    SFENCE();
    if (!ONEBAR && PF0 && pre) {
      read_a_s(nxt + HOFFR(0, 0), 0, 0);  // a0' (legal: proven at M)
      read_b_s(nxt + HOFFR(0, 2), 0, 0);  // b0'
    }
    if constexpr (ONEBAR)  // staging issued ~6 clusters ago: full slack
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();  // B
  }

  // epilogue: identical to the phased kernel's batched two-pass
  const u32 crow0 = qr + (lane >> 4) * 4;
  const u32 ccol0 = qc + (lane & 15);
  const int QMHR[4] = {0, 0, 1, 1};
  const int QNHR[4] = {0, 1, 1, 0};
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    float cv[2][4][4];
    if (!BETA0) {
#pragma unroll
      for (int m = 0; m < 2; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n) {
          long row = m0 + QMHR[q] * 128 + crow0 + m * 16;
          long col = n0 + QNHR[q] * 128 + ccol0 + n * 16;
          const float* cp = C + row * (long)N + col;
#pragma unroll
          for (int r = 0; r < 4; ++r) cv[m][n][r] = cp[(long)r * N];
        }
    }
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        long row = m0 + QMHR[q] * 128 + crow0 + m * 16;
        long col = n0 + QNHR[q] * 128 + ccol0 + n * 16;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = (BETA0 ? 0.0f : cv[m][n][r]) + acc[q][m][n][r];
          if (BF16OUT)
            Co[(row + r) * (long)N + col] = (bf16)v;
          else
            C[(row + r) * (long)N + col] = v;
        }
      }
  }
}
#undef MFMA_RS
#undef SFENCE
#undef HOFFR

// --------------------------- bf16, 256^2 tile, phased, 32x32x16 MFMA ----
// Same quadrant-per-phase pipeline as gemm_bt_bf16_acc_256p_k but on
// v_mfma_f32_32x32x16_bf16 (uarch ceiling 2382 TF vs 2075 for 16x16x32;
// half the MFMA instructions at 2x the flops each). Per quadrant a wave
// owns a 32 x 64 slice = 1 M-frag x 2 N-frags of 32x32, K=16 per MFMA
// (4 kk steps over BK=64). C/D map: col = lane&31,
// row = (reg&3) + 8*(reg>>2) + 4*(lane>>5).


__global__ void
__launch_bounds__(512)
gemm_bt_bf16_acc_256p32_k(float* __restrict__ C, const bf16* __restrict__ A,
                          const bf16* __restrict__ Bt, int M, int N, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
#define HOFF32(b, h) (((b) * 4u + (h)) * (128u * BKH * 2u))
  const u32 tid = threadIdx.x;
  const u32 lane = tid & 63, wave = tid >> 6;
  const u32 qr = (wave & 3) * 32;
  const u32 qc = (wave >> 2) * 64;

  const u32 ntm = (u32)(M / BM2), ntn = (u32)(N / BN2);
  u32 bid = xcd_swizzle(blockIdx.x, ntm * ntn);
  u32 tm, tn;
  if (ntm % 4 == 0 && ntn % 8 == 0) {
    supertile_map(bid, ntm, ntn, 4, 8, &tm, &tn);
  } else {
    tm = bid / ntn;
    tn = bid % ntn;
  }
  const long m0 = (long)tm * BM2, n0 = (long)tn * BN2;

  const u32 srow = tid >> 3, sslot = tid & 7;
  const u32 src_slot = sslot ^ (srow & 7);
  const int NT = K / BKH;

  auto stage_half = [&](u32 bufoff, int h, int t) {
    const long k0 = (long)t * BKH;
    const bf16* src = (h < 2)
        ? A + (m0 + (h & 1) * 128) * (long)K
        : Bt + (n0 + (h & 1) * 128) * (long)K;
    bf16* dst = (bf16*)(smem + bufoff + (u32)h * (128u * BKH * 2u));
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      u32 row = srow + it * 64;
      glds16(src + row * (long)K + k0 + src_slot * 8,
             dst + (row * 8u + sslot) * 8u);
    }
  };

  f32x16 acc[4][2] = {};   // [quadrant][n-frag]
  bf16x8 af[4], bfr[4][2];  // [kk] / [kk][n]

  const u32 arow_base = qr + (lane & 31);
  const u32 kgrp = lane >> 5;  // 0..1: 8-elem k sub-chunk within K=16

  auto read_a = [&](u32 off) {
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      u32 row = arow_base;
      u32 byte = row * 128u + (u32)kk * 32u + kgrp * 16u;
      byte ^= (row & 7) << 4;
      af[kk] = *(const bf16x8*)(smem + off + byte);
    }
  };
  auto read_b = [&](u32 off) {
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
#pragma unroll
      for (int n = 0; n < 2; ++n) {
        u32 row = qc + n * 32 + (lane & 31);
        u32 byte = row * 128u + (u32)kk * 32u + kgrp * 16u;
        byte ^= (row & 7) << 4;
        bfr[kk][n] = *(const bf16x8*)(smem + off + byte);
      }
  };

#define MFMA_QUAD32(q)                                                     \
  __builtin_amdgcn_s_setprio(1);                                          \
  _Pragma("unroll") for (int kk = 0; kk < 4; ++kk)                         \
      _Pragma("unroll") for (int n = 0; n < 2; ++n)                        \
          acc[q][n] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(             \
              af[kk], bfr[kk][n], acc[q][n], 0, 0, 0);                     \
  __builtin_amdgcn_s_setprio(0);

  stage_half(0u, 0, 0);
  stage_half(0u, 2, 0);
  stage_half(0u, 3, 0);
  stage_half(0u, 1, 0);
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  const u32 BUFSZ = 4u * 128u * BKH * 2u;
  for (int t = 0; t < NT; ++t) {
    const u32 cur = (u32)(t & 1) * BUFSZ, nxt = cur ^ BUFSZ;
    const bool pre = (t + 1 < NT);
    read_a(cur + HOFF32(0, 0));
    read_b(cur + HOFF32(0, 2));
    if (pre) stage_half(nxt, 0, t + 1);
    MFMA_QUAD32(0)
    if (pre) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else     asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    read_b(cur + HOFF32(0, 3));
    if (pre) stage_half(nxt, 2, t + 1);
    MFMA_QUAD32(1)
    if (pre) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else     asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    read_a(cur + HOFF32(0, 1));
    if (pre) stage_half(nxt, 3, t + 1);
    MFMA_QUAD32(2)
    __builtin_amdgcn_s_barrier();
    read_b(cur + HOFF32(0, 2));
    if (pre) stage_half(nxt, 1, t + 1);
    MFMA_QUAD32(3)
    if (pre) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  // epilogue: quadrant (mh,nh); C/D 32x32 map
  const int QMH32[4] = {0, 0, 1, 1};
  const int QNH32[4] = {0, 1, 1, 0};
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    float cv[2][16];
#pragma unroll
    for (int n = 0; n < 2; ++n)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        long row = m0 + QMH32[q] * 128 + qr + (r & 3) + 8 * (r >> 2) +
                   4 * (lane >> 5);
        long col = n0 + QNH32[q] * 128 + qc + n * 32 + (lane & 31);
        cv[n][r] = C[row * (long)N + col];
      }
#pragma unroll
    for (int n = 0; n < 2; ++n)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        long row = m0 + QMH32[q] * 128 + qr + (r & 3) + 8 * (r >> 2) +
                   4 * (lane >> 5);
        long col = n0 + QNH32[q] * 128 + qc + n * 32 + (lane & 31);
        C[row * (long)N + col] = cv[n][r] + acc[q][n][r];
      }
  }
}
#undef MFMA_QUAD32
#undef HOFF32

// ----------------------- bf16, 256^2 tile, 4 FAT waves (512 VGPR) --------
// Geometry reverse-engineered from hipBLASLt's winning gfx950 kernel for
// this shape class (Tensile name: MT256x256x64 MI16x16 MIWT8_8 WG32_8_1
// WS64 - profiles/README.md "Library comparison"): 256 threads = 4 waves
// at ONE wave per SIMD, each wave owning a 128x128 quadrant as an 8x8
// fragment grid held in a ~350-VGPR register tile (the unified 512-VGPR
// file is only addressable at occupancy 1). Payoff vs the 8-wave kernel:
// 16 ds_reads feed 64 MFMAs per kk (0.25 reads/MFMA vs 0.75) and there is
// no 2-wave/SIMD arbitration; latency is hidden by the wave's own ILP
// (64 independent MFMA chains) instead of a co-resident wave.

__global__ void
__launch_bounds__(256, 1)
__attribute__((amdgpu_waves_per_eu(1, 1)))
gemm_bt_bf16_acc_256w4_k(float* __restrict__ C, const bf16* __restrict__ A,
                         const bf16* __restrict__ Bt, int M, int N, int K,
                         u32 stm_, u32 stn_) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* As = (bf16*)smem;                        // [2][256*64]
  bf16* Bs = (bf16*)(smem + 2 * BM2 * BKH * 2);  // [2][256*64]

  const u32 tid = threadIdx.x;
  const u32 lane = tid & 63, wave = tid >> 6;
  const u32 wr = (wave >> 1) * 128, wc = (wave & 1) * 128;

  const u32 ntm = (u32)(M / BM2), ntn = (u32)(N / BN2);
  u32 bid = xcd_swizzle(blockIdx.x, ntm * ntn);
  u32 tm, tn;
  if (stm_ && ntm % stm_ == 0 && ntn % stn_ == 0) {
    supertile_map(bid, ntm, ntn, stm_, stn_, &tm, &tn);
  } else {
    tm = bid / ntn;
    tn = bid % ntn;
  }
  const long m0 = (long)tm * BM2, n0 = (long)tn * BN2;

  // staging: 256 rows x 8 slots per operand = 2048 chunks; 256 threads
  // -> 8 passes of 32 rows; same XOR-16 source swizzle as the 8-wave
  // kernels (identical LDS image, identical fragment reads).
  const u32 srow = tid >> 3, sslot = tid & 7;  // rows 0..31 per pass
  const u32 src_slot = sslot ^ (srow & 7);
  const int NT = K / BKH;

  // rolled (unroll 2): a fully unrolled 16-glds stage keeps 16 64-bit
  // addresses live and pushes the 512-VGPR allocation into in-loop
  // scratch spills; incremental pointers keep the register high-water
  // under the cap (VALU cost hides under the MFMA clusters).
  auto stage = [&](int buf, int t) {
    const long k0 = (long)t * BKH;
    const bf16* ga = A + (m0 + srow) * (long)K + k0 + src_slot * 8;
    const bf16* gb = Bt + (n0 + srow) * (long)K + k0 + src_slot * 8;
    bf16* la = As + buf * BM2 * BKH + (srow * 8u + sslot) * 8u;
    bf16* lb = Bs + buf * BM2 * BKH + (srow * 8u + sslot) * 8u;
#pragma unroll 2
    for (int it = 0; it < 8; ++it) {
      glds16(ga, la);
      glds16(gb, lb);
      ga += 32 * (long)K;
      gb += 32 * (long)K;
      la += 32 * 64;
      lb += 32 * 64;
    }
  };

  f32x4 acc[8][8] = {};  // 256 VGPRs: the 128x128 register tile
  const u32 arow_base = wr + (lane & 15);
  const u32 brow_base = wc + (lane & 15);
  const u32 kgrp = lane >> 4;

  stage(0, 0);
  __syncthreads();
  int cur = 0;
  for (int t = 0; t < NT; ++t) {
    if (t + 1 < NT) stage(cur ^ 1, t + 1);
    const char* a_base = (const char*)(As + cur * BM2 * BKH);
    const char* b_base = (const char*)(Bs + cur * BM2 * BKH);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[8], bf[8];
#pragma unroll
      for (int m = 0; m < 8; ++m) {
        u32 row = arow_base + m * 16;
        u32 byte = row * 128u + (u32)kk * 64u + kgrp * 16u;
        byte ^= (row & 7) << 4;
        af[m] = *(const bf16x8*)(a_base + byte);
      }
#pragma unroll
      for (int n = 0; n < 8; ++n) {
        u32 row = brow_base + n * 16;
        u32 byte = row * 128u + (u32)kk * 64u + kgrp * 16u;
        byte ^= (row & 7) << 4;
        bf[n] = *(const bf16x8*)(b_base + byte);
      }
#pragma unroll
      for (int m = 0; m < 8; ++m)
#pragma unroll
        for (int n = 0; n < 8; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[m], bf[n], acc[m][n], 0, 0, 0);
    }
    __syncthreads();
    cur ^= 1;
  }

  // epilogue: batched two-pass per fragment row
  const u32 crow0 = wr + (lane >> 4) * 4;
  const u32 ccol0 = wc + (lane & 15);
#pragma unroll
  for (int m = 0; m < 8; ++m) {
    float cv[8][4];
#pragma unroll
    for (int n = 0; n < 8; ++n) {
      long row = m0 + crow0 + m * 16;
      long col = n0 + ccol0 + n * 16;
      const float* cp = C + row * (long)N + col;
#pragma unroll
      for (int r = 0; r < 4; ++r) cv[n][r] = cp[(long)r * N];
    }
#pragma unroll
    for (int n = 0; n < 8; ++n) {
      long row = m0 + crow0 + m * 16;
      long col = n0 + ccol0 + n * 16;
      float* cp = C + row * (long)N + col;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        cp[(long)r * N] = cv[n][r] + acc[m][n][r];
    }
  }
}

// -------------------- fat-wave, software-pipelined (variant 13) ----------
// v12's deficit is the post-barrier fragment-read latency: at 1 wave/SIMD
// there is no co-resident wave to execute while this wave waits on its
// kk0 ds_reads. v13 pipelines WITHIN the wave: each tile's kk0 fragments
// are prefetched into registers under the PREVIOUS tile's kk1 MFMA
// cluster, so every MFMA cluster starts on proven-resident operands.
// Per tile: stage(t+1) | MFMA kk0 (prefetched) | reads kk1 | vmcnt+bar |
// MFMA kk1 | prefetch kk0(t+1) | bar. Both barriers land where the waits
// they imply are already satisfied. sched_barrier(0) fences pin the
// section order against scheduler hoisting (cf. the kk-split experiment).

__global__ void
__launch_bounds__(256, 1)
__attribute__((amdgpu_waves_per_eu(1, 1)))
gemm_bt_bf16_acc_256w4p_k(float* __restrict__ C,
                          const bf16* __restrict__ A,
                          const bf16* __restrict__ Bt, int M, int N, int K,
                          u32 stm_, u32 stn_) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* As = (bf16*)smem;                        // [2][256*64]
  bf16* Bs = (bf16*)(smem + 2 * BM2 * BKH * 2);  // [2][256*64]

  const u32 tid = threadIdx.x;
  const u32 lane = tid & 63, wave = tid >> 6;
  const u32 wr = (wave >> 1) * 128, wc = (wave & 1) * 128;

  const u32 ntm = (u32)(M / BM2), ntn = (u32)(N / BN2);
  u32 bid = xcd_swizzle(blockIdx.x, ntm * ntn);
  u32 tm, tn;
  if (stm_ && ntm % stm_ == 0 && ntn % stn_ == 0) {
    supertile_map(bid, ntm, ntn, stm_, stn_, &tm, &tn);
  } else {
    tm = bid / ntn;
    tn = bid % ntn;
  }
  const long m0 = (long)tm * BM2, n0 = (long)tn * BN2;

  const u32 srow = tid >> 3, sslot = tid & 7;
  const u32 src_slot = sslot ^ (srow & 7);
  const int NT = K / BKH;

  auto stage = [&](int buf, int t) {
    const long k0 = (long)t * BKH;
    const bf16* ga = A + (m0 + srow) * (long)K + k0 + src_slot * 8;
    const bf16* gb = Bt + (n0 + srow) * (long)K + k0 + src_slot * 8;
    bf16* la = As + buf * BM2 * BKH + (srow * 8u + sslot) * 8u;
    bf16* lb = Bs + buf * BM2 * BKH + (srow * 8u + sslot) * 8u;
#pragma unroll 2
    for (int it = 0; it < 8; ++it) {
      glds16(ga, la);
      glds16(gb, lb);
      ga += 32 * (long)K;
      gb += 32 * (long)K;
      la += 32 * 64;
      lb += 32 * 64;
    }
  };

  f32x4 acc[8][8] = {};
  bf16x8 af0[8], bf0[8];  // kk0 fragments (prefetched a tile ahead)
  bf16x8 af1[8], bf1[8];  // kk1 fragments
  const u32 arow_base = wr + (lane & 15);
  const u32 brow_base = wc + (lane & 15);
  const u32 kgrp = lane >> 4;

  auto read_kk = [&](int buf, int kk, bf16x8* afp, bf16x8* bfp) {
    const char* a_base = (const char*)(As + buf * BM2 * BKH);
    const char* b_base = (const char*)(Bs + buf * BM2 * BKH);
#pragma unroll
    for (int m = 0; m < 8; ++m) {
      u32 row = arow_base + m * 16;
      u32 byte = row * 128u + (u32)kk * 64u + kgrp * 16u;
      byte ^= (row & 7) << 4;
      afp[m] = *(const bf16x8*)(a_base + byte);
    }
#pragma unroll
    for (int n = 0; n < 8; ++n) {
      u32 row = brow_base + n * 16;
      u32 byte = row * 128u + (u32)kk * 64u + kgrp * 16u;
      byte ^= (row & 7) << 4;
      bfp[n] = *(const bf16x8*)(b_base + byte);
    }
  };

#define MFMA64(afp, bfp)                                                   \
  _Pragma("unroll") for (int m = 0; m < 8; ++m)                            \
      _Pragma("unroll") for (int n = 0; n < 8; ++n)                        \
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(             \
              afp[m], bfp[n], acc[m][n], 0, 0, 0);

  // prologue: stage tile 0, prove it, prefetch its kk0 fragments
  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  read_kk(0, 0, af0, bf0);

  // fine interleave directives: alternate {8 MFMA | 2 DS reads | 2 VMEM}
  // inside each half-tile region so the in-order wave keeps feeding the
  // MFMA pipe while its loads issue (single-wave: nothing else hides
  // issue bursts).
#define SCHED_MIX()                                                        \
  _Pragma("unroll") for (int g = 0; g < 8; ++g) {                          \
    __builtin_amdgcn_sched_group_barrier(0x008, 8, 0);                     \
    __builtin_amdgcn_sched_group_barrier(0x100, 2, 0);                     \
    __builtin_amdgcn_sched_group_barrier(0x020, 2, 0);                     \
  }

  int cur = 0;
  for (int t = 0; t < NT; ++t) {
    const bool pre = (t + 1 < NT);
    if (pre) stage(cur ^ 1, t + 1);             // 16 glds
    MFMA64(af0, bf0)                            // kk0, prefetched regs
    read_kk(cur, 1, af1, bf1);                  // kk1 (lands under kk0)
    SCHED_MIX()
    __builtin_amdgcn_sched_barrier(0);
    if (pre) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // stage in
    __builtin_amdgcn_s_barrier();               // nxt buffer readable
    MFMA64(af1, bf1)                            // kk1
    if (pre) read_kk(cur ^ 1, 0, af0, bf0);     // prefetch t+1 kk0
    SCHED_MIX()
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_barrier();               // cur safe to restage
    cur ^= 1;
  }
#undef SCHED_MIX
#undef MFMA64

  const u32 crow0 = wr + (lane >> 4) * 4;
  const u32 ccol0 = wc + (lane & 15);
#pragma unroll
  for (int m = 0; m < 8; ++m) {
    float cv[8][4];
#pragma unroll
    for (int n = 0; n < 8; ++n) {
      long row = m0 + crow0 + m * 16;
      long col = n0 + ccol0 + n * 16;
      const float* cp = C + row * (long)N + col;
#pragma unroll
      for (int r = 0; r < 4; ++r) cv[n][r] = cp[(long)r * N];
    }
#pragma unroll
    for (int n = 0; n < 8; ++n) {
      long row = m0 + crow0 + m * 16;
      long col = n0 + ccol0 + n * 16;
      float* cp = C + row * (long)N + col;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        cp[(long)r * N] = cv[n][r] + acc[m][n][r];
    }
  }
}

// ------------------------------------------------------------- launchers --

static int g_gemm_variant = 0;  // 0=auto, 1=simple 256, 2=phased 256, 3=128
extern "C" void set_gemm_variant(int v) { g_gemm_variant = v; }
unsigned g_st_m = 4, g_st_n = 4;  // supertile cluster dims (shared with gemm_mx8.hip)
extern "C" void set_supertile(int sm, int sn) {
  g_st_m = (unsigned)sm;
  g_st_n = (unsigned)sn;
}

extern "C" void launch_gemm_bt_bf16_acc(float* C, const void* A,
                                        const void* Bt, int M, int N, int K,
                                        hipStream_t stream);

// last-k fused variant: read fp32 partials from C (unless beta0), write
// the finished block STRAIGHT to bf16 Co (no fp32 store, no separate
// conversion pass). Only for the phased geometry; caller guards.
extern "C" int launch_gemm_bt_bf16_final(float* C, void* Co, const void* A,
                                         const void* Bt, int M, int N,
                                         int K, int beta0,
                                         hipStream_t stream) {
  int nb = (M / BM2) * (N / BN2);
  if (!(M % BM2 == 0 && N % BN2 == 0 && nb >= 192)) return 0;
  size_t lds = 2u * BM2 * BKH * 2 * 2;
  // ring becomes the default only after the measured A/B (set 14/15/16)
  bool ring = (g_gemm_variant == 14 || g_gemm_variant == 15 ||
               g_gemm_variant == 16);
  bool onebar = (g_gemm_variant == 16);
  if (ring && onebar && beta0)
    hipLaunchKernelGGL((gemm_bt_bf16_acc_256r_k<true, true, false, true>),
                       dim3((unsigned)nb), dim3(512), lds, stream, C,
                       (const bf16*)A, (const bf16*)Bt, M, N, K, g_st_m,
                       g_st_n, (bf16*)Co);
  else if (ring && onebar)
    hipLaunchKernelGGL((gemm_bt_bf16_acc_256r_k<false, true, false, true>),
                       dim3((unsigned)nb), dim3(512), lds, stream, C,
                       (const bf16*)A, (const bf16*)Bt, M, N, K, g_st_m,
                       g_st_n, (bf16*)Co);
  else if (ring && beta0)
    hipLaunchKernelGGL((gemm_bt_bf16_acc_256r_k<true, true>),
                       dim3((unsigned)nb), dim3(512), lds, stream, C,
                       (const bf16*)A, (const bf16*)Bt, M, N, K, g_st_m,
                       g_st_n, (bf16*)Co);
  else if (ring)
    hipLaunchKernelGGL((gemm_bt_bf16_acc_256r_k<false, true>),
                       dim3((unsigned)nb), dim3(512), lds, stream, C,
                       (const bf16*)A, (const bf16*)Bt, M, N, K, g_st_m,
                       g_st_n, (bf16*)Co);
  else if (beta0)
    hipLaunchKernelGGL((gemm_bt_bf16_acc_256p_k<false, true, true, true>),
                       dim3((unsigned)nb), dim3(512), lds, stream, C,
                       (const bf16*)A, (const bf16*)Bt, M, N, K, g_st_m,
                       g_st_n, (bf16*)Co);
  else
    hipLaunchKernelGGL((gemm_bt_bf16_acc_256p_k<false, true, false, true>),
                       dim3((unsigned)nb), dim3(512), lds, stream, C,
                       (const bf16*)A, (const bf16*)Bt, M, N, K, g_st_m,
                       g_st_n, (bf16*)Co);
  return 1;
}

extern "C" void launch_gemm_bt_bf16_acc_b0(float* C, const void* A,
                                           const void* Bt, int M, int N,
                                           int K, hipStream_t stream) {
  // beta=0 fast path (first k-contribution of a C block): phased kernel,
  // no C reads. Falls back to the generic path for small grids.
  int nb = (M / BM2) * (N / BN2);
  if (M % BM2 == 0 && N % BN2 == 0 && nb >= 192) {
    size_t lds = 2u * BM2 * BKH * 2 * 2;
    if (g_gemm_variant == 16)
      hipLaunchKernelGGL((gemm_bt_bf16_acc_256r_k<true, false, false, true>),
                         dim3((unsigned)nb), dim3(512), lds, stream, C,
                         (const bf16*)A, (const bf16*)Bt, M, N, K, g_st_m,
                         g_st_n);
    else if (g_gemm_variant == 14 || g_gemm_variant == 15)
      hipLaunchKernelGGL((gemm_bt_bf16_acc_256r_k<true, false>),
                         dim3((unsigned)nb), dim3(512), lds, stream, C,
                         (const bf16*)A, (const bf16*)Bt, M, N, K, g_st_m,
                         g_st_n);
    else
      hipLaunchKernelGGL((gemm_bt_bf16_acc_256p_k<false, true, true>),
                         dim3((unsigned)nb), dim3(512), lds, stream, C,
                         (const bf16*)A, (const bf16*)Bt, M, N, K, g_st_m,
                         g_st_n);
    return;
  }
  (void)hipMemsetAsync(C, 0, (size_t)M * N * sizeof(float), stream);
  launch_gemm_bt_bf16_acc(C, A, Bt, M, N, K, stream);
}

extern "C" void launch_gemm_bt_bf16_acc(float* C, const void* A,
                                        const void* Bt, int M, int N, int K,
                                        hipStream_t stream) {
  // 256^2 tile only when the grid still fills the chip (>= ~192 of 256
  // CUs at 1 block/CU); smaller problems keep the 128^2 tile (2 blocks/CU,
  // 4x the block count).
  int nb = (M / BM2) * (N / BN2);
  int v = g_gemm_variant;
  if (v == 0) v = (nb >= 512) ? 6 : (nb >= 192 ? 1 : 3);
  if (v != 3 && M % BM2 == 0 && N % BN2 == 0) {
    dim3 grid((unsigned)nb);
    size_t lds = 2u * BM2 * BKH * 2 * 2;  // 128 KiB
    if (v == 2)
      hipLaunchKernelGGL((gemm_bt_bf16_acc_256p_k<true>), grid, dim3(512),
                         lds, stream, C, (const bf16*)A, (const bf16*)Bt, M,
                         N, K, g_st_m, g_st_n);
    else if (v == 4)
      hipLaunchKernelGGL((gemm_bt_bf16_acc_256p_k<false>), grid, dim3(512),
                         lds, stream, C, (const bf16*)A, (const bf16*)Bt, M,
                         N, K, g_st_m, g_st_n);
    else if (v == 5)
      hipLaunchKernelGGL(gemm_bt_bf16_acc_256p32_k, grid, dim3(512), lds,
                         stream, C, (const bf16*)A, (const bf16*)Bt, M, N,
                         K);
    else if (v == 6)
      hipLaunchKernelGGL((gemm_bt_bf16_acc_256p_k<false, true>), grid,
                         dim3(512), lds, stream, C, (const bf16*)A,
                         (const bf16*)Bt, M, N, K, g_st_m, g_st_n);
    else if (v == 7)
      hipLaunchKernelGGL((gemm_bt_bf16_acc_256p_k<true, true>), grid,
                         dim3(512), lds, stream, C, (const bf16*)A,
                         (const bf16*)Bt, M, N, K, g_st_m, g_st_n);
    else if (v == 8)
      hipLaunchKernelGGL(
          (gemm_bt_bf16_acc_256p_k<false, true, false, false, true>), grid,
          dim3(512), lds, stream, C, (const bf16*)A, (const bf16*)Bt, M, N,
          K, g_st_m, g_st_n);
    else if (v == 9)
      hipLaunchKernelGGL(
          (gemm_bt_bf16_acc_256p_k<false, false, false, false, true>), grid,
          dim3(512), lds, stream, C, (const bf16*)A, (const bf16*)Bt, M, N,
          K, g_st_m, g_st_n);
    else if (v == 10)
      hipLaunchKernelGGL(
          (gemm_bt_bf16_acc_256p_k<false, true, false, false, false, true>),
          grid, dim3(512), lds, stream, C, (const bf16*)A, (const bf16*)Bt,
          M, N, K, g_st_m, g_st_n);
    else if (v == 14)
      hipLaunchKernelGGL((gemm_bt_bf16_acc_256r_k<false, false, true>),
                         grid, dim3(512), lds, stream, C, (const bf16*)A,
                         (const bf16*)Bt, M, N, K, g_st_m, g_st_n);
    else if (v == 15)
      hipLaunchKernelGGL((gemm_bt_bf16_acc_256r_k<false, false, false>),
                         grid, dim3(512), lds, stream, C, (const bf16*)A,
                         (const bf16*)Bt, M, N, K, g_st_m, g_st_n);
    else if (v == 16)
      hipLaunchKernelGGL(
          (gemm_bt_bf16_acc_256r_k<false, false, false, true>), grid,
          dim3(512), lds, stream, C, (const bf16*)A, (const bf16*)Bt, M, N,
          K, g_st_m, g_st_n);
    else if (v == 12)
      hipLaunchKernelGGL(gemm_bt_bf16_acc_256w4_k, grid, dim3(256), lds,
                         stream, C, (const bf16*)A, (const bf16*)Bt, M, N,
                         K, g_st_m, g_st_n);
    else if (v == 13)
      hipLaunchKernelGGL(gemm_bt_bf16_acc_256w4p_k, grid, dim3(256), lds,
                         stream, C, (const bf16*)A, (const bf16*)Bt, M, N,
                         K, g_st_m, g_st_n);
    else
      hipLaunchKernelGGL(gemm_bt_bf16_acc_256_k, grid, dim3(512), lds,
                         stream, C, (const bf16*)A, (const bf16*)Bt, M, N,
                         K);
    return;
  }
  dim3 grid((M / BM) * (N / BN));
  size_t lds = 2u * BM * BKH * 2 * 2;  // 2 bufs x (A+B) x 16 KiB
  hipLaunchKernelGGL(gemm_bt_bf16_acc_k, grid, dim3(256), lds, stream, C,
                     (const bf16*)A, (const bf16*)Bt, M, N, K);
}

static int g_f32_variant = 0;  // 0 = 32x32x2 (default), 1 = 16x16x4 256-tile
extern "C" void set_f32_variant(int v) { g_f32_variant = v; }

extern "C" void launch_gemm_bt_f32_acc(float* C, const float* A,
                                       const float* Bt, int M, int N, int K,
                                       hipStream_t stream) {
  // 256^2 tile when the shape allows and the grid still fills the chip;
  // fall back to the 128^2 kernel for small/ragged-tile problems.
  int nb = (M / BM2) * (N / BN2);
  if (M % BM2 == 0 && N % BN2 == 0 && nb >= 192) {
    size_t lds = 2u * BM2 * BKF * 4 * 2;  // 128 KiB
    if (g_f32_variant == 1)
      hipLaunchKernelGGL(gemm_bt_f32_acc_256_k, dim3((unsigned)nb),
                         dim3(512), lds, stream, C, A, Bt, M, N, K);
    else
      hipLaunchKernelGGL(gemm_bt_f32_acc_256x32_k, dim3((unsigned)nb),
                         dim3(512), lds, stream, C, A, Bt, M, N, K);
    return;
  }
  dim3 grid((M / BM) * (N / BN));
  size_t lds = 2u * BM * BKF * 4 * 2;
  hipLaunchKernelGGL(gemm_bt_f32_acc_k, grid, dim3(256), lds, stream, C, A,
                     Bt, M, N, K);
}
