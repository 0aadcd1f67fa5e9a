// Standalone MX-fp8 MFMA layout probe (not part of the extension).
// Empirically maps operand/scale lane layouts of
// v_mfma_scale_f32_32x32x64_f8f6f4.
// Build: hipcc --offload-arch=gfx950 -O2 scripts/mx_probe.hip -o scripts/mx_probe
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <cstring>

typedef int v8i __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

__global__ void probe_k(const unsigned char* __restrict__ A,
                        const unsigned char* __restrict__ B,
                        float* __restrict__ C, int sa, int sb) {
  int lane = threadIdx.x;
  v8i a = ((const v8i*)A)[lane];
  v8i b = ((const v8i*)B)[lane];
  f32x16 c = {};
  // mode>=1000: per-lane scale probes. 1000+x: sa = 127 + (lane&1),
  // sb=127 -> which C rows double tells us which lanes feed which rows.
  int sa_eff = sa, sb_eff = sb;
  if (sa >= 1000) {
    int mode = sa - 1000;
    if (mode == 0) sa_eff = 127 + (lane & 1);          // by lane parity
    else if (mode == 1) sa_eff = 127 + ((lane >> 5) & 1);  // by k-half
    else if (mode == 2) sa_eff = (lane == 3) ? 128 : 127;  // single lane
    else if (mode == 1000) sa_eff = (lane == 35) ? 128 : 127;
    else sa_eff = 127;
  }
  c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, c, 0, 0, 0,
                                                      sa_eff, 0, sb_eff);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    int col = lane & 31;
    C[row * 32 + col] = c[r];
  }
}

static unsigned char f2e4m3(float f) {
  if (f == 0.0f) return 0;
  unsigned s = f < 0 ? 0x80 : 0;
  float a = fabsf(f);
  int e = (int)floorf(log2f(a));
  if (e < -6) return s;
  float m = a / exp2f((float)e) - 1.0f;
  int mant = (int)roundf(m * 8.0f);
  if (mant == 8) { mant = 0; e += 1; }
  int be = e + 7;
  if (be <= 0 || be > 15) return s;
  return (unsigned char)(s | (be << 3) | mant);
}

unsigned char hA[64 * 32], hB[64 * 32];
float hC[32 * 32];
unsigned char *dA, *dB;
float* dC;

static void run(int sa, int sb) {
  (void)hipMemcpy(dA, hA, sizeof(hA), hipMemcpyHostToDevice);
  (void)hipMemcpy(dB, hB, sizeof(hB), hipMemcpyHostToDevice);
  (void)hipMemset(dC, 0, sizeof(hC));
  hipLaunchKernelGGL(probe_k, dim3(1), dim3(64), 0, 0, dA, dB, dC, sa, sb);
  (void)hipDeviceSynchronize();
  (void)hipMemcpy(hC, dC, sizeof(hC), hipMemcpyDeviceToHost);
}

int main() {
  (void)hipMalloc(&dA, sizeof(hA));
  (void)hipMalloc(&dB, sizeof(hB));
  (void)hipMalloc(&dC, sizeof(hC));

  // ---- test 1: all ones, scale calibration (observed: (127,127)->2^14
  // per product; find S0 with (S0,S0) -> x1)
  unsigned char one = f2e4m3(1.0f);
  memset(hA, one, sizeof(hA));
  memset(hB, one, sizeof(hB));
  for (int s : {110, 113, 120, 125, 127}) {
    run(s, s);
    printf("ones sa=sb=%d: C[0][0]=%g (want 64)\n", s, hC[0]);
  }
  run(113, 127);
  printf("ones sa=113 sb=127: C[0][0]=%g\n", hC[0]);

  // ---- test 2: single nonzero in A at lane 0 byte 0; B all ones
  memset(hA, 0, sizeof(hA));
  hA[0] = one;  // lane 0, byte 0
  run(127, 127);
  printf("single A lane0 byte0: nonzero C entries:\n");
  for (int i = 0; i < 32; ++i)
    for (int j = 0; j < 32; ++j)
      if (hC[i * 32 + j] != 0.0f)
        printf("  C[%d][%d] = %g\n", i, j, hC[i * 32 + j]);

  // ---- test 3: single nonzero at lane 5, byte 17
  memset(hA, 0, sizeof(hA));
  hA[5 * 32 + 17] = one;
  run(127, 127);
  printf("single A lane5 byte17: nonzero C entries:\n");
  for (int i = 0; i < 32; ++i)
    for (int j = 0; j < 32; ++j)
      if (hC[i * 32 + j] != 0.0f)
        printf("  C[%d][%d] = %g\n", i, j, hC[i * 32 + j]);

  // ---- test 4: single nonzero at lane 40, byte 3
  memset(hA, 0, sizeof(hA));
  hA[40 * 32 + 3] = one;
  run(127, 127);
  printf("single A lane40 byte3: nonzero C entries:\n");
  for (int i = 0; i < 32; ++i)
    for (int j = 0; j < 32; ++j)
      if (hC[i * 32 + j] != 0.0f)
        printf("  C[%d][%d] = %g\n", i, j, hC[i * 32 + j]);

  // ---- test 5: A all ones; single nonzero in B at lane 9 byte 30
  memset(hA, one, sizeof(hA));
  memset(hB, 0, sizeof(hB));
  hB[9 * 32 + 30] = one;
  run(127, 127);
  printf("single B lane9 byte30: nonzero rows at col 9: C[0][9]=%g\n",
         hC[9]);

  // ---- test 6: k-position cross-check: A nonzero ONLY at (lane0,byte7),
  // B nonzero ONLY at (lane0,byte7) and (lane0,byte8): C[0][0] nonzero
  // iff both occupy the same k -> byte index IS the k offset within the
  // lane's 32-k block.
  memset(hA, 0, sizeof(hA));
  memset(hB, 0, sizeof(hB));
  hA[7] = one;
  hB[7] = one;
  run(127, 127);
  printf("k-match byte7/byte7: C[0][0]=%g (expect nonzero)\n", hC[0]);
  hB[7] = 0;
  hB[8] = one;
  run(127, 127);
  printf("k-mismatch byte7/byte8: C[0][0]=%g (expect 0)\n", hC[0]);

  // ---- test 7: per-lane scale independence (lane 0 gets sa+1)
  //      -> C row 0 should be 1.5x other rows (half its k doubled)
  memset(hA, one, sizeof(hA));
  memset(hB, one, sizeof(hB));
  {
    // per-lane scale needs a per-lane value; reuse kernel by passing
    // lane-dependent sa via a small variant: here emulate with sa s.t.
    // lane-uniform (documented limitation of this probe); per-lane
    // behaviour is exercised in the real kernel's unit test instead.
  }

  // ---- test 8: full random check with calibrated scales
  srand(7);
  const float vals[7] = {0.f, 0.5f, 1.f, 1.5f, -0.5f, -1.f, 2.f};
  float fA[32][64], fB[64][32];
  for (int i = 0; i < 32; ++i)
    for (int k = 0; k < 64; ++k) fA[i][k] = vals[rand() % 7];
  for (int k = 0; k < 64; ++k)
    for (int j = 0; j < 32; ++j) fB[k][j] = vals[rand() % 7];
  for (int l = 0; l < 64; ++l) {
    int row = l & 31, kh = l >> 5;
    for (int kk = 0; kk < 32; ++kk) {
      hA[l * 32 + kk] = f2e4m3(fA[row][kh * 32 + kk]);
      hB[l * 32 + kk] = f2e4m3(fB[kh * 32 + kk][row]);
    }
  }
  {
    run(127, 127);
    double maxerr = 0;
    int shown = 0;
    for (int i = 0; i < 32; ++i)
      for (int j = 0; j < 32; ++j) {
        double ref = 0;
        for (int k = 0; k < 64; ++k) ref += (double)fA[i][k] * fB[k][j];
        double e = fabs(hC[i * 32 + j] - ref);
        if (e > 1e-3 && shown < 8) {
          printf("  mismatch C[%d][%d]: got %g want %g\n", i, j,
                 hC[i * 32 + j], ref);
          ++shown;
        }
        if (e > maxerr) maxerr = e;
      }
    printf("random check s0=127: max err %.6f %s\n", maxerr,
           maxerr < 1e-3 ? "PASS" : "FAIL");
  }

  // negatives test: A all -1, B all 1 -> C = -64
  memset(hA, f2e4m3(-1.0f), sizeof(hA));
  memset(hB, f2e4m3(1.0f), sizeof(hB));
  run(127, 127);
  printf("neg ones: C[0][0]=%g (want -64)\n", hC[0]);

  // 0.5 test: A all 0.5, B all 0.5 -> C = 16
  memset(hA, f2e4m3(0.5f), sizeof(hA));
  memset(hB, f2e4m3(0.5f), sizeof(hB));
  run(127, 127);
  printf("halves: C[0][0]=%g (want 16)\n", hC[0]);

  // 1.5 test: A all 1.5, B all 1.0 -> 96
  memset(hA, f2e4m3(1.5f), sizeof(hA));
  memset(hB, f2e4m3(1.0f), sizeof(hB));
  run(127, 127);
  printf("1.5s: C[0][0]=%g (want 96)\n", hC[0]);

  // 2.0 test
  memset(hA, f2e4m3(2.0f), sizeof(hA));
  run(127, 127);
  printf("2.0s: C[0][0]=%g (want 128)\n", hC[0]);

  // per-lane scale probes (A=B=1.0)
  memset(hA, f2e4m3(1.0f), sizeof(hA));
  memset(hB, f2e4m3(1.0f), sizeof(hB));
  run(1000, 127);  // sa = 127 + (lane&1)
  printf("perlane parity: C[0][0]=%g C[1][0]=%g C[2][0]=%g C[3][0]=%g\n",
         hC[0], hC[32], hC[64], hC[96]);
  run(1001, 127);  // sa = 127 + k-half
  printf("perlane khalf: C[r][c] all should be 96: C[0][0]=%g C[5][9]=%g\n",
         hC[0], hC[5 * 32 + 9]);
  run(1002, 127);  // only lane 3 -> rows/k it feeds doubled
  printf("single lane3 x2:\n");
  for (int i = 0; i < 32; ++i)
    if (hC[i * 32] != 64.0f) printf("  C[%d][0]=%g\n", i, hC[i * 32]);

  // DECISIVE k-half pairing probe: payload only in k-half 0 (lanes<32),
  // scale doubled only on lane 3 (a k-half-0 lane). If lane-l scale pairs
  // with lane-l data (same k-half): C[3][0] = 64; if the halves are
  // swapped: C[3][0] = 32 (the doubled scale lands on the zero half).
  memset(hA, 0, sizeof(hA));
  memset(hB, 0, sizeof(hB));
  for (int l = 0; l < 32; ++l)
    for (int kk = 0; kk < 32; ++kk) {
      hA[l * 32 + kk] = f2e4m3(1.0f);
      hB[l * 32 + kk] = f2e4m3(1.0f);
    }
  run(1002, 127);
  printf("khalf pairing: C[3][0]=%g (64=same-lane, 32=swapped), "
         "C[4][0]=%g (ctrl, want 32)\n", hC[3 * 32], hC[4 * 32]);
  // and the mirror: scale doubled on lane 35 (k-half-1 lane), payload
  // still only in half 0: same-lane -> no effect (32); swapped -> 64.
  // reuse mode: sa>=2000 -> lane==35 doubled
  run(2000, 127);
  printf("khalf mirror: C[3][0]=%g (32=same-lane, 64=swapped)\n",
         hC[3 * 32]);

  // ---- decode-table sweep: A filled with byte v, B = 1.0 (0x38);
  // hardware value of byte v = C[0][0] / 64.
  memset(hB, 0x38, sizeof(hB));
  printf("decode table (byte: hw_value | e4m3_value):\n");
  for (int v = 0x30; v < 0x48; v += 1) {
    memset(hA, v, sizeof(hA));
    run(120, 120);
    float e4 = 0.f;
    {
      int be = (v >> 4) & 0xF, m = v & 7;
      e4 = be ? exp2f((float)(be - 7)) * (1 + m / 8.0f)
              : exp2f(-6.f) * (m / 8.0f);
    }
    printf("  0x%02x: %g | %g\n", v, hC[0] / 64.0f, e4);
  }

  // random payload through the bare MFMA vs host dequant model
  srand(123);
  for (int i = 0; i < (int)sizeof(hA); ++i) hA[i] = rand() & 0x7F;
  for (int i = 0; i < (int)sizeof(hB); ++i) hB[i] = rand() & 0x7F;
  run(127, 127);
  {
    double maxerr = 0;
    for (int i = 0; i < 32; ++i)
      for (int j = 0; j < 32; ++j) {
        double ref = 0;
        for (int kk = 0; kk < 64; ++kk) {
          int la = (kk >= 32 ? 32 : 0) + i, ka = kk & 31;
          int lb = (kk >= 32 ? 32 : 0) + j, kb2 = kk & 31;
          auto dec = [](unsigned char v) -> double {
            int s = v >> 7, be = (v >> 3) & 0xF, m = v & 7;
            double val = be ? exp2((double)(be - 7)) * (1 + m / 8.0)
                            : exp2(-6.0) * (m / 8.0);
            return s ? -val : val;
          };
          ref += dec(hA[la * 32 + ka]) * dec(hB[lb * 32 + kb2]);
        }
        double e = fabs(hC[i * 32 + j] - ref);
        if (e > maxerr) maxerr = e;
      }
    printf("bare-mfma random-payload vs dequant model: max err %.6f %s\n",
           maxerr, maxerr < 1e-2 ? "PASS" : "FAIL");
  }
  return 0;
}
