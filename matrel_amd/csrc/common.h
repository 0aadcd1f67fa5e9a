// Common device helpers for the matrel_amd CDNA4 (gfx950) kernels.
// Target: MI355X only - wave64, 4 SIMD-32/CU, 160 KiB LDS, 8 XCDs.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define NXCD 8

typedef __bf16 bf16;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef unsigned int u32;
typedef unsigned long long u64;

// ---- async global->LDS copy (16 B per lane; dest wave-uniform base +
// lane*16, source per-lane). See cdna_hip_programming.md §5.
__device__ __forceinline__ void glds16(const void* gptr, void* lds_ptr) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) u32*)gptr,
      (__attribute__((address_space(3))) u32*)lds_ptr, 16, 0, 0);
}

__device__ __forceinline__ void glds4(const void* gptr, void* lds_ptr) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) u32*)gptr,
      (__attribute__((address_space(3))) u32*)lds_ptr, 4, 0, 0);
}

// ---- XCD-aware block swizzle (bijective form; cdna_hip_programming.md §5):
// consecutive output tiles land on the same XCD's L2 so neighbouring tiles
// re-hit shared operand panels.
__device__ __forceinline__ u32 xcd_swizzle(u32 bid, u32 nwg) {
  if (nwg < 2 * NXCD) return bid;
  u32 xcd = bid % NXCD, q = nwg / NXCD, r = nwg % NXCD;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + bid / NXCD;
}

// ---- supertile decomposition: virtual id -> (tm, tn) such that ids
// 0..SM*SN-1 form a 2D SM x SN cluster of output tiles (the ~32 tiles
// co-resident on one XCD then share SM A-strips + SN B-strips instead of
// 32 disjoint ones: the GEMM at >=8k is HBM-bound, so strip reuse through
// the XCD's L2/LLC is the main lever past ~1 TF/s... per-GPU TFLOP).
// Requires ntm % SM == 0 && ntn % SN == 0 (caller guards).
__device__ __forceinline__ void supertile_map(u32 id, u32 ntm, u32 ntn,
                                              u32 SM, u32 SN, u32* tm,
                                              u32* tn) {
  u32 per = SM * SN;
  u32 st = id / per, w = id % per;
  u32 stn_cnt = ntn / SN;
  u32 stm = st / stn_cnt, stn = st % stn_cnt;
  *tm = stm * SM + w / SN;
  *tn = stn * SN + w % SN;
}

// ---- ordered-int encoding for float atomic max/min
__device__ __forceinline__ int f32_order(float x) {
  int i = __float_as_int(x);
  return i >= 0 ? i : ~i;
}
__device__ __forceinline__ float f32_unorder(int i) {
  return __int_as_float(i >= 0 ? i : ~i);
}

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      throw std::runtime_error(std::string("HIP error: ") +                 \
                               hipGetErrorString(_e));                      \
    }                                                                       \
  } while (0)
