// bf16 -> MX-fp8 conversion: per (row, 32-wide k-block) compute the e8m0
// shared scale (2^(floor(log2 maxabs) - 7): normalized max in [128,256),
// inside e4m3's 448 range) and quantize the 32 values to e4m3 (RNE).
// Feeds csrc/gemm_mx8.hip.
#include "common.h"

typedef __bf16 bf16x8v __attribute__((ext_vector_type(8)));

__device__ __forceinline__ unsigned char f32_to_e4m3(float f) {
  // OCP e4m3fn, round-to-nearest-even, saturate to +-448
  unsigned s = (__float_as_uint(f) >> 31) << 7;
  float a = fabsf(f);
  if (a != a) return (unsigned char)(s | 0x7F);  // NaN
  if (a >= 448.0f) return (unsigned char)(s | 0x7E);  // saturate to 448
  if (a < 0.87890625e-2f * 0.125f) {  // < 2^-10 -> subnormal/zero region
    // subnormals: m/8 * 2^-6
    int m = (int)rintf(a * 512.0f);  // a / 2^-9
    if (m > 7) m = 7;
    return (unsigned char)(s | m);
  }
  int e = (int)floorf(log2f(a));
  float mant = a / exp2f((float)e) - 1.0f;  // [0,1)
  int m = (int)rintf(mant * 8.0f);
  if (m == 8) { m = 0; e += 1; }
  int be = e + 7;
  if (be < 1) {  // subnormal
    int ms = (int)rintf(a * 512.0f);
    if (ms > 7) ms = 7;
    return (unsigned char)(s | ms);
  }
  if (be > 15) return (unsigned char)(s | 0x7E);
  return (unsigned char)(s | (be << 3) | m);
}

// one thread per (row, kblock): 32 bf16 in, 32 e4m3 + 1 e8m0 out
__global__ void mx8_convert_k(unsigned char* __restrict__ out8,
                              unsigned char* __restrict__ sc,
                              const bf16* __restrict__ in, long rows,
                              long cols) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long kb = cols / 32;
  if (idx >= rows * kb) return;
  long r = idx / kb, b = idx - r * kb;
  const bf16* src = in + r * cols + b * 32;
  float v[32];
  float mx = 0.0f;
#pragma unroll
  for (int i = 0; i < 32; i += 8) {
    bf16x8v x = *(const bf16x8v*)(src + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = (float)x[j];
      v[i + j] = f;
      float a = fabsf(f);
      if (a > mx) mx = a;
    }
  }
  int e;
  // Scale layout (round 2): when cols % 128 == 0 (every engine path -
  // blocks are % 256), scales are PACKED DWORDS sc32[k_tile][row] with
  // byte b128 = block 4*k_tile + b128 - the BK=128 GEMM kernels then
  // load ONE coalesced dword per fragment row per K-tile instead of 4
  // strided bytes (24 strided byte loads/tile was a measured stall).
  // Otherwise the legacy block-major byte layout sc[block][row] feeds
  // the BK=64 kernel.
  long sidx = (cols % 128 == 0)
      ? ((b >> 2) * rows + r) * 4 + (b & 3)
      : b * rows + r;
  if (mx == 0.0f) {
    e = 0;
    sc[sidx] = 127;  // scale 1.0; all-zero payload
  } else {
    e = (int)floorf(log2f(mx)) - 7;  // normalized max in [128, 256)
    if (e < -127) e = -127;
    if (e > 127) e = 127;
    sc[sidx] = (unsigned char)(e + 127);
  }
  float inv = exp2f((float)-e);
  // pack 32 e4m3 bytes into 8 dwords and store as two 16-B vectors
  unsigned int words[8];
#pragma unroll
  for (int w = 0; w < 8; ++w) {
    unsigned int x = 0;
#pragma unroll
    for (int j = 0; j < 4; ++j)
      x |= (unsigned int)f32_to_e4m3(v[w * 4 + j] * inv) << (8 * j);
    words[w] = x;
  }
  typedef unsigned int u32x4 __attribute__((ext_vector_type(4)));
  u32x4* dst = (u32x4*)(out8 + r * cols + b * 32);
  dst[0] = u32x4{words[0], words[1], words[2], words[3]};
  dst[1] = u32x4{words[4], words[5], words[6], words[7]};
}

extern "C" void launch_mx8_convert(void* out8, void* sc, const void* in,
                                   long rows, long cols, hipStream_t s) {
  long n = rows * (cols / 32);
  long blocks = (n + 255) / 256;
  hipLaunchKernelGGL(mx8_convert_k, dim3((unsigned)blocks), dim3(256), 0, s,
                     (unsigned char*)out8, (unsigned char*)sc,
                     (const bf16*)in, rows, cols);
}

// ---- bf16 -> MX-fp4 (e2m1 + e8m0 per-32-block scales) -------------------
// Layout verified by scripts/mx4_probe.hip: the fp8 byte map at NIBBLE
// granularity (payload = 16 bytes/lane in the LOW v4i half of the
// mfma_scale operand). Memory layout: nibble k of a row-block packed 2
// per byte in original-k order ([rows][cols/2] bytes); scales use the
// SAME packed-dword layout as fp8 ([cols/128][rows] int32), so the
// SUMMA wire/caches treat both precisions uniformly. Scale exponent:
// e = floor(log2 max) - 1 normalizes the block max into [2,4) - inside
// e2m1's 6.0 range with no clipping (half a bit of headroom traded for
// clip-free RNE).

__device__ __forceinline__ unsigned char f32_to_e2m1(float f) {
  unsigned s = (__float_as_uint(f) >> 31) << 3;
  float a = fabsf(f);
  if (a != a) return (unsigned char)(s | 0x7);  // NaN -> max mag
  // representable magnitudes: 0, .5, 1, 1.5, 2, 3, 4, 6 (codes 0..7)
  // round-to-nearest-even on the midpoints
  unsigned char m;
  if (a <= 0.25f) m = 0;       // 0.25 -> even 0
  else if (a < 0.75f) m = 1;
  else if (a <= 1.25f) m = 2;       // 0.75..1.25 -> 1.0 (1.25->even 1.0)
  else if (a < 1.75f) m = 3;
  else if (a <= 2.5f) m = 4;        // 2.5 -> even 2.0
  else if (a < 3.5f) m = 5;
  else if (a <= 5.0f) m = 6;        // 5.0 -> even 4.0
  else m = 7;
  return (unsigned char)(s | m);
}

__global__ void mx4_convert_k(unsigned char* __restrict__ out4,
                              unsigned char* __restrict__ sc,
                              const bf16* __restrict__ in, long rows,
                              long cols) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long kb = cols / 32;
  if (idx >= rows * kb) return;
  long r = idx / kb, b = idx - r * kb;
  const bf16* src = in + r * cols + b * 32;
  float v[32];
  float mx = 0.0f;
#pragma unroll
  for (int i = 0; i < 32; i += 8) {
    bf16x8v x = *(const bf16x8v*)(src + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = (float)x[j];
      v[i + j] = f;
      float a = fabsf(f);
      if (a > mx) mx = a;
    }
  }
  int e;
  // packed-dword scale layout (cols % 128 == 0 enforced by the binding)
  long sidx = ((b >> 2) * rows + r) * 4 + (b & 3);
  if (mx == 0.0f) {
    e = 0;
    sc[sidx] = 127;
  } else {
    e = (int)floorf(log2f(mx)) - 1;  // block max lands in [2, 4)
    if (e < -127) e = -127;
    if (e > 127) e = 127;
    sc[sidx] = (unsigned char)(e + 127);
  }
  float inv = exp2f((float)-e);
  // pack 32 nibbles into 16 bytes (4 dwords), original-k order
  unsigned int words[4];
#pragma unroll
  for (int w = 0; w < 4; ++w) {
    unsigned int x = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      x |= (unsigned int)f32_to_e2m1(v[w * 8 + j] * inv) << (4 * j);
    words[w] = x;
  }
  typedef unsigned int u32x4c __attribute__((ext_vector_type(4)));
  *(u32x4c*)(out4 + r * (cols / 2) + b * 16) =
      u32x4c{words[0], words[1], words[2], words[3]};
}

extern "C" void launch_mx4_convert(void* out4, void* sc, const void* in,
                                   long rows, long cols, hipStream_t s) {
  long n = rows * (cols / 32);
  long blocks = (n + 255) / 256;
  hipLaunchKernelGGL(mx4_convert_k, dim3((unsigned)blocks), dim3(256), 0,
                     s, (unsigned char*)out4, (unsigned char*)sc,
                     (const bf16*)in, rows, cols);
}

// ---- bf16 -> MX-fp6 (e2m3 + e8m0 per-32-block scales) -------------------
// Layout verified by scripts/mx6_probe.hip: LSB-first 6-bit fields
// (32 per block = 24 bytes), per-SOURCE-LANE scale coverage like fp4.
// Scale exponent: e = floor(log2 max) - 2 normalizes the block max into
// [4, 8) (e2m3 tops out at 7.5: only a max element in (7.25, 8) rounds
// onto 7.5 - <= 6% on that one element). Payload: [rows][cols*3/4]
// bytes; scales: packed-dword [cols/128][rows] int32 (same as fp8/fp4).

__device__ __forceinline__ unsigned char f32_to_e2m3(float f) {
  unsigned s = (__float_as_uint(f) >> 31) << 5;
  float a = fabsf(f);
  if (a != a) return (unsigned char)(s | 0x1F);
  // grid: subnormals m/8 (m 0..7), then (1+m/8)*2^(e-1), e 1..3; RNE
  // equivalently: q = round(a * 8) in [0, 15] covers codes 0..15
  // (values 0..1.875 step .125); above that the step doubles per octave
  unsigned c;
  if (a < 2.0f) {
    int q = (int)rintf(a * 8.0f);
    if (q > 15) q = 16;  // 1.9375.. -> 2.0
    c = (unsigned)q;
    if (q == 16) c = 0x10;
  } else if (a < 4.0f) {
    int q = (int)rintf(a * 4.0f);  // step .25: codes 0x10..0x17
    if (q > 15) q = 16;
    c = (unsigned)(0x10 + (q - 8));
    if (q == 16) c = 0x18;
  } else {
    int q = (int)rintf(a * 2.0f);  // step .5: codes 0x18..0x1F
    if (q > 15) q = 15;            // saturate at 7.5
    if (q < 8) q = 8;
    c = (unsigned)(0x18 + (q - 8));
  }
  return (unsigned char)(s | c);
}

__global__ void mx6_convert_k(unsigned char* __restrict__ out6,
                              unsigned char* __restrict__ sc,
                              const bf16* __restrict__ in, long rows,
                              long cols) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long kb = cols / 32;
  if (idx >= rows * kb) return;
  long r = idx / kb, b = idx - r * kb;
  const bf16* src = in + r * cols + b * 32;
  float v[32];
  float mx = 0.0f;
#pragma unroll
  for (int i = 0; i < 32; i += 8) {
    bf16x8v x = *(const bf16x8v*)(src + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = (float)x[j];
      v[i + j] = f;
      float a = fabsf(f);
      if (a > mx) mx = a;
    }
  }
  int e;
  long sidx = ((b >> 2) * rows + r) * 4 + (b & 3);
  if (mx == 0.0f) {
    e = 0;
    sc[sidx] = 127;
  } else {
    e = (int)floorf(log2f(mx)) - 2;  // block max lands in [4, 8)
    if (e < -127) e = -127;
    if (e > 127) e = 127;
    sc[sidx] = (unsigned char)(e + 127);
  }
  float inv = exp2f((float)-e);
  // pack 32 x 6-bit fields LSB-first into 24 bytes (6 dwords)
  unsigned int w[6] = {0, 0, 0, 0, 0, 0};
#pragma unroll
  for (int f = 0; f < 32; ++f) {
    unsigned int code = f32_to_e2m3(v[f] * inv);
    int bit = f * 6, d = bit >> 5, off = bit & 31;
    w[d] |= code << off;
    if (off > 26) w[d + 1] |= code >> (32 - off);
  }
  // SPLIT-PLANE layout: L plane row-major [rows][cols/32 x 16B]
  // (payload bytes 0-15 of each 24-B block = mfma regs 0-3), then H
  // plane BLOCK-major [cols/32][rows][8B] (bytes 16-23 = regs 4-5).
  // The split keeps every GEMM LDS read power-of-2; the H plane's
  // block-major order makes each K-tile's H bytes one contiguous
  // 2-KiB-per-block run, so the GEMM's staging DMA issues full-line
  // coalesced requests (a row-major H plane fragments into 32-B
  // segments and measured -25% end to end) and the b64 reads land on a
  // conflict-free 2-dword lane stride.
  typedef unsigned int u32x4c __attribute__((ext_vector_type(4)));
  typedef unsigned int u32x2c __attribute__((ext_vector_type(2)));
  *(u32x4c*)(out6 + r * (cols / 2) + b * 16) =
      u32x4c{w[0], w[1], w[2], w[3]};
  *(u32x2c*)(out6 + rows * (cols / 2) + (b * rows + r) * 8) =
      u32x2c{w[4], w[5]};
}

extern "C" void launch_mx6_convert(void* out6, void* sc, const void* in,
                                   long rows, long cols, hipStream_t s) {
  long n = rows * (cols / 32);
  long blocks = (n + 255) / 256;
  hipLaunchKernelGGL(mx6_convert_k, dim3((unsigned)blocks), dim3(256), 0,
                     s, (unsigned char*)out6, (unsigned char*)sc,
                     (const bf16*)in, rows, cols);
}
