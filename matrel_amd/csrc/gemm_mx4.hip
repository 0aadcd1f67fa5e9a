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

__global__ void gemm_bt_mx4_acc_k128r3(float* __restrict__,
                                        const unsigned char* __restrict__,
                                        const unsigned char* __restrict__,
                                        const unsigned char* __restrict__,
                                        const unsigned char* __restrict__,
                                        int, int, int, int, u32, u32);
__global__ void gemm_bt_mx4_acc_k256(float* __restrict__,
                                     const unsigned char* __restrict__,
                                     const unsigned char* __restrict__,
                                     const unsigned char* __restrict__,
                                     const unsigned char* __restrict__,
                                     int, int, int, int, u32, u32);

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

// ------------------------- triple-buffered ring (overlapped staging) ----
// The double-buffered kernel above serializes staging with compute: the
// compiler's memory legalizer inserts s_waitcnt vmcnt(0) before the
// fragment ds_reads (it cannot prove the in-flight global_load_lds
// writes target the OTHER buffer), so every K-tile pays
// mfma-time + staging-time. Measured decomposition at 16384^3
// (scripts/mx4_floor_ab.py): full 2865 TF vs staging-removed 4358 TF.
// This variant stages tile t+2 during tile t through a 3-buffer LDS
// ring (96 KiB) with v6-style counted waits (gemm.hip): per tile,
//   reads (v6 phase order: ds_reads BEFORE this tile's DMA issue) ->
//   [scales t+1 (6 dwords), stage t+2 (4 glds)] -> mfmas ->
//   s_waitcnt vmcnt(4) (proves S_{t+1}+G_{t+1}, leaves G_{t+2} in
//   flight - never drains to 0 in the main loop) -> raw s_barrier.
// The wait is UNCONDITIONAL straight-line code (a branchy wait splits
// the block and the legalizer's conservative merge re-adds a vmcnt(0)
// drain) and carries no "memory" clobber (the legalizer treats
// clobbering asm as a memory access and drains before it). Scale banks
// alternate by tile parity (2x unroll, literal banks): register copies
// or %2 indexing both re-serialize (the copies make the legalizer
// drain for the pending scale loads; dynamic indexing spills).
// Buffer identity rotates through three pointer pairs.
// The end-of-tile barrier separates tile t's reads of its buffer from
// tile t+1's DMA into that same buffer (t+3 = t mod 3). Sync-structure
// change => race-screened (multi-run bit-exactness vs the
// double-buffered kernel); MATREL_MX4_V=1 keeps the old kernel.
__global__ void
__launch_bounds__(512)
gemm_bt_mx4_acc_k128r3(float* __restrict__ C,
                       const unsigned char* __restrict__ A,
                       const unsigned char* __restrict__ Asc,
                       const unsigned char* __restrict__ Bt,
                       const unsigned char* __restrict__ Bsc, int M,
                       int N, int K, int beta0, u32 stm_, u32 stn_) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  unsigned char* As3 = (unsigned char*)smem;              // [3][16 KiB]
  unsigned char* Bs3 = (unsigned char*)(smem + 3 * 16384);

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
  const long KB2 = K / 2;

  const u32 srow = tid >> 2;
  const u32 sslot = tid & 3;
  const u32 src_slot = sslot ^ ((srow >> 2) & 3);

  // per-thread staging source pointers as STABLE loop-carried registers
  // STEPPED by += 64 per tile (no per-call offset addition): an addr
  // temp computed per tile gets register-reused as a ds_read dest, and
  // the waitcnt pass then protects the pending DMA's address regs with
  // a vmcnt(0) drain - the serialization this kernel exists to remove.
  // per-thread staging source pointers as STABLE loop-carried registers
  // STEPPED by += 64 per tile (no per-call offset addition): an addr
  // temp computed per tile gets register-reused as a ds_read dest, and
  // the waitcnt pass then protects the pending DMA's address regs with
  // a vmcnt(0) drain in the middle of the tile.
  const unsigned char* sa0 = A + (m0 + srow) * KB2 + src_slot * 16;
  const unsigned char* sa1 = A + (m0 + srow + 128) * KB2 + src_slot * 16;
  const unsigned char* sb0 = Bt + (n0 + srow) * KB2 + src_slot * 16;
  const unsigned char* sb1 = Bt + (n0 + srow + 128) * KB2 + src_slot * 16;

  f32x16q acc[4][2] = {};
  const u32 arow = wm * 128 + (lane & 31);
  const u32 brow = wn * 64 + (lane & 31);
  const u32 kh = lane >> 5;

  auto loadsc = [&](int t, int* sa_w, int* sb_w) {
#pragma unroll
    for (int m = 0; m < 4; ++m)
      sa_w[m] = ((const int*)Asc)[(long)t * M + m0 + arow + m * 32];
#pragma unroll
    for (int n = 0; n < 2; ++n)
      sb_w[n] = ((const int*)Bsc)[(long)t * N + n0 + brow + n * 32];
  };

  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  const int NT = K / 128;
  // buffer ring as three wave-uniform 32-bit offsets (SGPRs; 64-bit
  // pointer pairs cost 12 VGPRs and spilled the loop)
  u32 oc = 0, on = 16384, os2 = 32768;
  int sa[4], sb[2];

  // prologue: [G0, G1] in flight; scales are loaded per tile (no
  // prefetch bank: bank rotation needs copies or a 2x unroll, and both
  // re-serialize - copies make the legalizer drain for the pending
  // loads, the unroll spills; a per-tile scale load costs only its L2
  // latency and its counted wait leaves the DMA ring in flight).
  auto stage_to = [&](u32 buf, long k0b) {
    glds16(sa0 + k0b, As3 + buf + srow * 64u + sslot * 16u);
    glds16(sa1 + k0b, As3 + buf + (srow + 128) * 64u + sslot * 16u);
    glds16(sb0 + k0b, Bs3 + buf + srow * 64u + sslot * 16u);
    glds16(sb1 + k0b, Bs3 + buf + (srow + 128) * 64u + sslot * 16u);
  };
  stage_to(oc, 0);
  if (NT > 1) stage_to(on, 64);
  // pre-advance to tile 2: in-loop staging reads the pointers directly
  sa0 += 128; sa1 += 128; sb0 += 128; sb1 += 128;
  asm volatile("s_waitcnt vmcnt(4)");  // tile 0 proven (G1 in flight)
  __builtin_amdgcn_s_barrier();

#define MX4R3_PHASE(KK, ISSUE)                                              \
  {                                                                         \
    v8i4 af[4], bf[2];                                                      \
    _Pragma("unroll")                                                       \
    for (int m = 0; m < 4; ++m) {                                           \
      u32 row = arow + m * 32;                                              \
      u32 byte = row * 64u +                                                \
                 (((u32)(KK) * 2u + kh) ^ ((row >> 2) & 3)) * 16u;          \
      af[m] = v4_from16((const char*)As3 + oc + byte);                      \
    }                                                                       \
    _Pragma("unroll")                                                       \
    for (int n = 0; n < 2; ++n) {                                           \
      u32 row = brow + n * 32;                                              \
      u32 byte = row * 64u +                                                \
                 (((u32)(KK) * 2u + kh) ^ ((row >> 2) & 3)) * 16u;          \
      bf[n] = v4_from16((const char*)Bs3 + oc + byte);                      \
    }                                                                       \
    ISSUE;                                                                  \
    _Pragma("unroll")                                                       \
    for (int m = 0; m < 4; ++m)                                             \
      _Pragma("unroll")                                                     \
      for (int n = 0; n < 2; ++n)                                           \
        acc[m][n] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(        \
            af[m], bf[n], acc[m][n], 4, 4, 0,                               \
            (sa[m] >> (8 * ((KK) * 2 + (int)kh))) & 0xFF, 0,                \
            (sb[n] >> (8 * ((KK) * 2 + (int)kh))) & 0xFF);                  \
  }

  loadsc(0, sa, sb);
  // tile body: reads+mfmas first; ALL new memory issue at tile END so
  // every glds -> ds_read path crosses the barrier (the structure the
  // legalizer accepts without a mid-tile vmcnt(0) drain; scales BEFORE
  // glds so next tile's scale extraction is a counted wait). The
  // pointer steps on the backedge still cost one vmcnt(0) there -
  // but by then the staging DMA has had the whole mfma cluster to
  // complete, which is the overlap this kernel exists to create.
  for (int t = 0; t < NT; ++t) {
    MX4R3_PHASE(0, {});
    __builtin_amdgcn_sched_barrier(0);
    MX4R3_PHASE(1, {});
    __builtin_amdgcn_sched_barrier(0);
    if (t + 1 < NT) loadsc(t + 1, sa, sb);
    if (t + 2 < NT) {
      glds16(sa0, As3 + os2 + srow * 64u + sslot * 16u);
      glds16(sa1, As3 + os2 + (srow + 128) * 64u + sslot * 16u);
      glds16(sb0, Bs3 + os2 + srow * 64u + sslot * 16u);
      glds16(sb1, Bs3 + os2 + (srow + 128) * 64u + sslot * 16u);
    }
    __builtin_amdgcn_sched_barrier(0);
    asm volatile("s_waitcnt vmcnt(10)");
    __builtin_amdgcn_s_barrier();
    u32 tmp_ = oc; oc = on; on = os2; os2 = tmp_;
    sa0 += 64; sa1 += 64; sb0 += 64; sb1 += 64;
  }
#undef MX4R3_PHASE

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

// ------------------------- K-tile 256 (halved per-tile sync) ----------
// Same proven k128 LDS layout, but each buffer holds TWO k128 sub-tiles
// (2 x 2 x 32 KiB = 128 KiB): one barrier + one staging drain per 256
// of K instead of per 128, halving the per-K sync overhead (the
// request-throughput study bounds staging itself; this trims the fixed
// cost 'a'). MATREL_MX4_V=2. Race-screened bit-exact vs the default.
__global__ void
__launch_bounds__(512)
gemm_bt_mx4_acc_k256(float* __restrict__ C,
                     const unsigned char* __restrict__ A,
                     const unsigned char* __restrict__ Asc,
                     const unsigned char* __restrict__ Bt,
                     const unsigned char* __restrict__ Bsc, int M,
                     int N, int K, int beta0, u32 stm_, u32 stn_) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  unsigned char* As = (unsigned char*)smem;  // [2][2][16 KiB]
  unsigned char* Bs = (unsigned char*)(smem + 2 * 32768);

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
  const long KB2 = K / 2;

  const u32 srow = tid >> 2;
  const u32 sslot = tid & 3;
  const u32 src_slot = sslot ^ ((srow >> 2) & 3);

  // stage one K-256 tile = two k128 sub-tiles per operand (8 glds)
  auto stage = [&](int buf, int t) {
    const long k0b = (long)t * 128;  // K-tile of 256 = 128 packed bytes
#pragma unroll
    for (int sub = 0; sub < 2; ++sub)
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        u32 row = srow + half * 128;
        glds16(A + (m0 + row) * KB2 + k0b + sub * 64 + src_slot * 16,
               As + buf * 32768u + sub * 16384u + row * 64u + sslot * 16u);
        glds16(Bt + (n0 + row) * KB2 + k0b + sub * 64 + src_slot * 16,
               Bs + buf * 32768u + sub * 16384u + row * 64u + sslot * 16u);
      }
  };

  f32x16q acc[4][2] = {};
  const u32 arow = wm * 128 + (lane & 31);
  const u32 brow = wn * 64 + (lane & 31);
  const u32 kh = lane >> 5;

  // packed-dword scales: K-256 tile spans TWO scale words per row
  auto load_scales = [&](int t, int* sa_w, int* sb_w) {
#pragma unroll
    for (int q = 0; q < 2; ++q) {
#pragma unroll
      for (int m = 0; m < 4; ++m)
        sa_w[q * 4 + m] =
            ((const int*)Asc)[(long)(2 * t + q) * M + m0 + arow + m * 32];
#pragma unroll
      for (int n = 0; n < 2; ++n)
        sb_w[q * 2 + n] =
            ((const int*)Bsc)[(long)(2 * t + q) * N + n0 + brow + n * 32];
    }
  };

  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  int sa_cur[8], sb_cur[4], sa_nxt[8], sb_nxt[4];
  load_scales(0, sa_cur, sb_cur);
  stage(0, 0);
  __syncthreads();
  int cur = 0;
  const int NT = K / 256;
  for (int t = 0; t < NT; ++t) {
    const bool pre = (t + 1 < NT);
    if (pre) {
      stage(cur ^ 1, t + 1);
      load_scales(t + 1, sa_nxt, sb_nxt);
    }
    const unsigned char* a_base = As + cur * 32768u;
    const unsigned char* b_base = Bs + cur * 32768u;
#pragma unroll
    for (int kq = 0; kq < 4; ++kq) {  // sub-tile = kq>>1, inner kk = kq&1
      const u32 sub = (u32)kq >> 1, kk = (u32)kq & 1;
      v8i4 af[4], bf[2];
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        u32 row = arow + m * 32;
        u32 byte = sub * 16384u + row * 64u +
                   ((kk * 2u + kh) ^ ((row >> 2) & 3)) * 16u;
        af[m] = v4_from16((const char*)a_base + byte);
      }
#pragma unroll
      for (int n = 0; n < 2; ++n) {
        u32 row = brow + n * 32;
        u32 byte = sub * 16384u + row * 64u +
                   ((kk * 2u + kh) ^ ((row >> 2) & 3)) * 16u;
        bf[n] = v4_from16((const char*)b_base + byte);
      }
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 2; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              af[m], bf[n], acc[m][n], 4, 4, 0,
              (sa_cur[sub * 4 + m] >> (8 * ((int)kk * 2 + (int)kh))) & 0xFF,
              0,
              (sb_cur[sub * 2 + n] >> (8 * ((int)kk * 2 + (int)kh))) & 0xFF);
    }
    __syncthreads();
#pragma unroll
    for (int m = 0; m < 8; ++m) sa_cur[m] = sa_nxt[m];
#pragma unroll
    for (int n = 0; n < 4; ++n) sb_cur[n] = sb_nxt[n];
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
