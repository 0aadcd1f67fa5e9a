// Python bindings for the matrel_amd CDNA4 kernels (torch extension).
// Host-side only: tensor checks, workspace allocation, kernel dispatch.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <stdexcept>

// launchers from the .hip translation units
extern "C" {
void launch_gemm_bt_bf16_acc(float*, const void*, const void*, int, int,
                             int, hipStream_t);
void launch_gemm_bt_bf16_acc_b0(float*, const void*, const void*, int, int,
                                int, hipStream_t);
int launch_gemm_bt_bf16_final(float*, void*, const void*, const void*, int,
                              int, int, int, hipStream_t);
void launch_gemm_bt_mx8_acc(float*, const void*, const void*, const void*,
                            const void*, int, int, int, int, hipStream_t);
void launch_mx8_convert(void*, void*, const void*, long, long, hipStream_t);
void launch_mx4_convert(void*, void*, const void*, long, long, hipStream_t);
void launch_mx6_convert(void*, void*, const void*, long, long, hipStream_t);
void launch_gemm_bt_mx6_acc(float*, const void*, const void*, const void*,
                            const void*, int, int, int, int, hipStream_t);
void launch_gemm_bt_mx4_acc(float*, const void*, const void*, const void*,
                            const void*, int, int, int, int, hipStream_t);
void launch_gemm_bt_f32_acc(float*, const float*, const float*, int, int,
                            int, hipStream_t);
void set_gemm_variant(int);
void set_mx8_variant(int);
void set_f32_variant(int);
void set_supertile(int, int);
void launch_ewise_bf16(int, void*, const void*, const void*, long,
                       hipStream_t);
void launch_ewise_f32(int, float*, const float*, const float*, long,
                      hipStream_t);
void launch_ewise_scalar_bf16(int, void*, const void*, float, long,
                              hipStream_t);
void launch_ewise_scalar_f32(int, float*, const float*, float, long,
                             hipStream_t);
void launch_mask_value_bf16(void*, const void*, float, float, int, long,
                            hipStream_t);
void launch_mask_value_f32(float*, const float*, float, float, int, long,
                           hipStream_t);
void launch_mask_rows_cols_bf16(void*, const void*, const unsigned char*,
                                const unsigned char*, long, long,
                                hipStream_t);
void launch_mask_rows_cols_f32(float*, const float*, const unsigned char*,
                               const unsigned char*, long, long,
                               hipStream_t);
void launch_outer_f32(int, float*, const float*, const float*, long, long,
                      hipStream_t);
void launch_mask_isin_bf16(void*, const void*, const float*, int, long,
                           hipStream_t);
void launch_mask_isin_f32(float*, const float*, const float*, int, long,
                          hipStream_t);
void launch_reduce_full_bf16(float*, float*, const void*, long, int,
                             hipStream_t);
void launch_reduce_full_f32(float*, float*, const float*, long, int,
                            hipStream_t);
void launch_reduce_rows_bf16(float*, const void*, long, long, int,
                             hipStream_t);
void launch_reduce_rows_f32(float*, const float*, long, long, int,
                            hipStream_t);
void launch_reduce_cols_bf16(float*, float*, const void*, long, long, int,
                             hipStream_t);
void launch_reduce_cols_f32(float*, float*, const float*, long, long, int,
                            hipStream_t);
int reduce_cols_nchunks(long, long);
void launch_trace_bf16(float*, float*, const void*, long, long, hipStream_t);
void launch_trace_f32(float*, float*, const float*, long, long, hipStream_t);
void launch_transpose_bf16(void*, const void*, long, long, hipStream_t);
void launch_spmm_csr_acc_bf16(float*, const int*, const int*, const void*,
                              const void*, long, long, hipStream_t);
void launch_spmm_csr_acc_f32(float*, const int*, const int*, const float*,
                             const float*, long, long, hipStream_t);
int launch_spgemm_csr_csr_bf16(float*, const int*, const int*, const void*,
                               const int*, const int*, const void*, long,
                               long, hipStream_t);
int launch_spgemm_csr_csr_f32(float*, const int*, const int*, const float*,
                              const int*, const int*, const float*, long,
                              long, hipStream_t);
void launch_transpose_f32(float*, const float*, long, long, hipStream_t);
void launch_ewise_transpose_bf16(int, void*, const void*, const void*, long,
                                 long, hipStream_t);
void launch_ewise_transpose_f32(int, float*, const float*, const float*,
                                long, long, hipStream_t);
}

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_2d(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.dim() == 2, name, " must be 2D");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

bool is_bf16(const torch::Tensor& t) {
  return t.scalar_type() == torch::kBFloat16;
}

// ---- GEMM: C(f32) += A @ Bt^T; shapes must be padded to the kernel tile
// (M,N % 128, K % 64 for bf16 / % 32 for f32). Python side pads.
void gemm_bt_acc_impl(torch::Tensor& C, torch::Tensor& A,
                      torch::Tensor& Bt, bool beta0);

void gemm_bt_acc(torch::Tensor C, torch::Tensor A, torch::Tensor Bt) {
  gemm_bt_acc_impl(C, A, Bt, false);
}

void gemm_bt_acc_beta0(torch::Tensor C, torch::Tensor A, torch::Tensor Bt) {
  // C may be UNINITIALIZED (torch.empty): the kernel overwrites it.
  gemm_bt_acc_impl(C, A, Bt, true);
}

void gemm_bt_acc_impl(torch::Tensor& C, torch::Tensor& A, torch::Tensor& Bt,
                      bool beta0) {
  check_2d(C, "C");
  check_2d(A, "A");
  check_2d(Bt, "Bt");
  TORCH_CHECK(C.scalar_type() == torch::kFloat32, "C must be fp32");
  int M = (int)A.size(0), K = (int)A.size(1), N = (int)Bt.size(0);
  TORCH_CHECK(Bt.size(1) == K, "K mismatch");
  TORCH_CHECK(C.size(0) == M && C.size(1) == N, "C shape mismatch");
  TORCH_CHECK(M % 128 == 0 && N % 128 == 0, "M,N must be multiples of 128");
  if (is_bf16(A)) {
    TORCH_CHECK(is_bf16(Bt), "A/Bt dtype mismatch");
    TORCH_CHECK(K % 64 == 0, "K must be a multiple of 64 (bf16)");
    if (beta0)
      launch_gemm_bt_bf16_acc_b0(C.data_ptr<float>(), A.data_ptr(),
                                 Bt.data_ptr(), M, N, K, cur_stream());
    else
      launch_gemm_bt_bf16_acc(C.data_ptr<float>(), A.data_ptr(),
                              Bt.data_ptr(), M, N, K, cur_stream());
  } else {
    if (beta0) C.zero_();
    TORCH_CHECK(A.scalar_type() == torch::kFloat32, "A must be bf16/fp32");
    TORCH_CHECK(K % 32 == 0, "K must be a multiple of 32 (fp32)");
    launch_gemm_bt_f32_acc(C.data_ptr<float>(), A.data_ptr<float>(),
                           Bt.data_ptr<float>(), M, N, K, cur_stream());
  }
}

torch::Tensor ewise(int64_t op, torch::Tensor a, torch::Tensor b) {
  check_2d(a, "a");
  check_2d(b, "b");
  TORCH_CHECK(a.sizes() == b.sizes() && a.scalar_type() == b.scalar_type());
  auto out = torch::empty_like(a);
  long n = a.numel();
  if (is_bf16(a))
    launch_ewise_bf16((int)op, out.data_ptr(), a.data_ptr(), b.data_ptr(), n,
                      cur_stream());
  else
    launch_ewise_f32((int)op, out.data_ptr<float>(), a.data_ptr<float>(),
                     b.data_ptr<float>(), n, cur_stream());
  return out;
}

torch::Tensor ewise_scalar(int64_t op, torch::Tensor a, double s) {
  check_2d(a, "a");
  auto out = torch::empty_like(a);
  long n = a.numel();
  if (is_bf16(a))
    launch_ewise_scalar_bf16((int)op, out.data_ptr(), a.data_ptr(), (float)s,
                             n, cur_stream());
  else
    launch_ewise_scalar_f32((int)op, out.data_ptr<float>(),
                            a.data_ptr<float>(), (float)s, n, cur_stream());
  return out;
}

torch::Tensor mask_value(torch::Tensor a, double lo, double hi, bool inside) {
  check_2d(a, "a");
  auto out = torch::empty_like(a);
  long n = a.numel();
  if (is_bf16(a))
    launch_mask_value_bf16(out.data_ptr(), a.data_ptr(), (float)lo,
                           (float)hi, inside ? 1 : 0, n, cur_stream());
  else
    launch_mask_value_f32(out.data_ptr<float>(), a.data_ptr<float>(),
                          (float)lo, (float)hi, inside ? 1 : 0, n,
                          cur_stream());
  return out;
}

torch::Tensor mask_rows_cols(torch::Tensor a,
                             c10::optional<torch::Tensor> rm,
                             c10::optional<torch::Tensor> cm) {
  check_2d(a, "a");
  auto out = torch::empty_like(a);
  const unsigned char* rmp = nullptr;
  const unsigned char* cmp = nullptr;
  torch::Tensor rmt, cmt;
  if (rm.has_value()) {
    rmt = rm->to(a.device(), torch::kUInt8).contiguous();
    TORCH_CHECK(rmt.numel() == a.size(0), "row mask length");
    rmp = rmt.data_ptr<unsigned char>();
  }
  if (cm.has_value()) {
    cmt = cm->to(a.device(), torch::kUInt8).contiguous();
    TORCH_CHECK(cmt.numel() == a.size(1), "col mask length");
    cmp = cmt.data_ptr<unsigned char>();
  }
  if (is_bf16(a))
    launch_mask_rows_cols_bf16(out.data_ptr(), a.data_ptr(), rmp, cmp,
                               a.size(0), a.size(1), cur_stream());
  else
    launch_mask_rows_cols_f32(out.data_ptr<float>(), a.data_ptr<float>(),
                              rmp, cmp, a.size(0), a.size(1), cur_stream());
  return out;
}

torch::Tensor outer(int64_t op, torch::Tensor u, torch::Tensor v) {
  TORCH_CHECK(u.is_cuda() && v.is_cuda());
  auto uf = u.to(torch::kFloat32).contiguous().reshape(-1);
  auto vf = v.to(torch::kFloat32).contiguous().reshape(-1);
  long m = uf.numel(), n = vf.numel();
  auto out = torch::empty({m, n}, uf.options());
  launch_outer_f32((int)op, out.data_ptr<float>(), uf.data_ptr<float>(),
                   vf.data_ptr<float>(), m, n, cur_stream());
  return out;
}

torch::Tensor mask_isin(torch::Tensor a, torch::Tensor needles_sorted) {
  check_2d(a, "a");
  auto nd = needles_sorted.to(a.device(), torch::kFloat32).contiguous();
  auto out = torch::empty_like(a);
  long n = a.numel();
  if (is_bf16(a))
    launch_mask_isin_bf16(out.data_ptr(), a.data_ptr(),
                          nd.data_ptr<float>(), (int)nd.numel(), n,
                          cur_stream());
  else
    launch_mask_isin_f32(out.data_ptr<float>(), a.data_ptr<float>(),
                         nd.data_ptr<float>(), (int)nd.numel(), n,
                         cur_stream());
  return out;
}

torch::Tensor reduce2d(torch::Tensor a, int64_t kind, int64_t axis) {
  check_2d(a, "a");
  long rows = a.size(0), cols = a.size(1);
  auto opts = torch::TensorOptions()
                  .dtype(torch::kFloat32)
                  .device(a.device());
  if (axis < 0) {
    auto out = torch::empty({1}, opts);
    auto partials = torch::empty({1024}, opts);
    if (is_bf16(a))
      launch_reduce_full_bf16(out.data_ptr<float>(),
                              partials.data_ptr<float>(), a.data_ptr(),
                              rows * cols, (int)kind, cur_stream());
    else
      launch_reduce_full_f32(out.data_ptr<float>(),
                             partials.data_ptr<float>(),
                             a.data_ptr<float>(), rows * cols, (int)kind,
                             cur_stream());
    return out;
  }
  if (axis == 1) {
    auto out = torch::empty({rows}, opts);
    if (is_bf16(a))
      launch_reduce_rows_bf16(out.data_ptr<float>(), a.data_ptr(), rows,
                              cols, (int)kind, cur_stream());
    else
      launch_reduce_rows_f32(out.data_ptr<float>(), a.data_ptr<float>(),
                             rows, cols, (int)kind, cur_stream());
    return out;
  }
  auto out = torch::empty({cols}, opts);
  int nc = reduce_cols_nchunks(rows, cols);
  auto partials = torch::empty({(long)nc * cols}, opts);
  if (is_bf16(a))
    launch_reduce_cols_bf16(out.data_ptr<float>(),
                            partials.data_ptr<float>(), a.data_ptr(), rows,
                            cols, (int)kind, cur_stream());
  else
    launch_reduce_cols_f32(out.data_ptr<float>(),
                           partials.data_ptr<float>(), a.data_ptr<float>(),
                           rows, cols, (int)kind, cur_stream());
  return out;
}

torch::Tensor trace(torch::Tensor a) {
  check_2d(a, "a");
  long n = std::min(a.size(0), a.size(1));
  auto opts =
      torch::TensorOptions().dtype(torch::kFloat32).device(a.device());
  auto out = torch::empty({1}, opts);
  auto partials = torch::empty({1024}, opts);
  if (is_bf16(a))
    launch_trace_bf16(out.data_ptr<float>(), partials.data_ptr<float>(),
                      a.data_ptr(), n, a.size(1), cur_stream());
  else
    launch_trace_f32(out.data_ptr<float>(), partials.data_ptr<float>(),
                     a.data_ptr<float>(), n, a.size(1), cur_stream());
  return out;
}

// C(f32, MxN) += CSR(M x K) @ B(K x N)
void spmm_csr_acc(torch::Tensor C, torch::Tensor crow, torch::Tensor cols,
                  torch::Tensor vals, torch::Tensor B) {
  check_2d(C, "C");
  check_2d(B, "B");
  TORCH_CHECK(C.scalar_type() == torch::kFloat32, "C must be fp32");
  TORCH_CHECK(crow.scalar_type() == torch::kInt32 &&
              cols.scalar_type() == torch::kInt32, "indices must be int32");
  long M = C.size(0), N = C.size(1);
  TORCH_CHECK(B.size(1) == N, "N mismatch");
  TORCH_CHECK(crow.numel() == M + 1, "crow length");
  auto cr = crow.contiguous();
  auto cl = cols.contiguous();
  auto vl = vals.contiguous();
  if (vals.scalar_type() == torch::kBFloat16) {
    TORCH_CHECK(is_bf16(B), "vals/B dtype mismatch");
    launch_spmm_csr_acc_bf16(C.data_ptr<float>(), cr.data_ptr<int>(),
                             cl.data_ptr<int>(), vl.data_ptr(), B.data_ptr(),
                             M, N, cur_stream());
  } else {
    launch_spmm_csr_acc_f32(C.data_ptr<float>(), cr.data_ptr<int>(),
                            cl.data_ptr<int>(), vl.data_ptr<float>(),
                            B.data_ptr<float>(), M, N, cur_stream());
  }
}

// C(f32, MxN) += A_csr(M x K) @ B_csr(K x N) - sparse x sparse block
// product (LDS ds_add_f32 accumulator row; spgemm.hip). Returns false
// when N exceeds the 64 KiB LDS row (caller falls back to densified B).
bool spgemm_csr_csr_acc(torch::Tensor C, torch::Tensor crowA,
                        torch::Tensor colsA, torch::Tensor valsA,
                        torch::Tensor crowB, torch::Tensor colsB,
                        torch::Tensor valsB) {
  check_2d(C, "C");
  TORCH_CHECK(C.scalar_type() == torch::kFloat32, "C must be fp32");
  TORCH_CHECK(crowA.scalar_type() == torch::kInt32 &&
              colsA.scalar_type() == torch::kInt32 &&
              crowB.scalar_type() == torch::kInt32 &&
              colsB.scalar_type() == torch::kInt32, "indices must be int32");
  TORCH_CHECK(valsA.scalar_type() == valsB.scalar_type(),
              "operand value dtypes must match");
  long M = C.size(0), N = C.size(1);
  TORCH_CHECK(crowA.numel() == M + 1, "crowA length");
  auto crA = crowA.contiguous(), clA = colsA.contiguous();
  auto vlA = valsA.contiguous();
  auto crB = crowB.contiguous(), clB = colsB.contiguous();
  auto vlB = valsB.contiguous();
  int ok;
  if (valsA.scalar_type() == torch::kBFloat16) {
    ok = launch_spgemm_csr_csr_bf16(
        C.data_ptr<float>(), crA.data_ptr<int>(), clA.data_ptr<int>(),
        vlA.data_ptr(), crB.data_ptr<int>(), clB.data_ptr<int>(),
        vlB.data_ptr(), M, N, cur_stream());
  } else {
    TORCH_CHECK(valsA.scalar_type() == torch::kFloat32, "bf16/fp32 only");
    ok = launch_spgemm_csr_csr_f32(
        C.data_ptr<float>(), crA.data_ptr<int>(), clA.data_ptr<int>(),
        vlA.data_ptr<float>(), crB.data_ptr<int>(), clB.data_ptr<int>(),
        vlB.data_ptr<float>(), M, N, cur_stream());
  }
  return ok != 0;
}

// bf16 -> (e4m3 bytes, e8m0 scales [rows][cols/32])
std::vector<torch::Tensor> mx8_convert(torch::Tensor a) {
  check_2d(a, "a");
  TORCH_CHECK(is_bf16(a), "mx8_convert needs bf16");
  long rows = a.size(0), cols = a.size(1);
  TORCH_CHECK(cols % 32 == 0, "cols must be a multiple of 32");
  auto opts = torch::TensorOptions().dtype(torch::kUInt8).device(a.device());
  auto out8 = torch::empty({rows, cols}, opts);
  // packed-dword scale layout when cols % 128 == 0 (see mxconvert.hip)
  auto sc = (cols % 128 == 0)
      ? torch::empty({cols / 128, rows},
                     torch::TensorOptions().dtype(torch::kInt32)
                         .device(a.device()))
      : torch::empty({cols / 32, rows}, opts);
  launch_mx8_convert(out8.data_ptr(), sc.data_ptr(), a.data_ptr(), rows,
                     cols, cur_stream());
  return {out8, sc};
}

// bf16 -> MX-fp4: packed e2m1 nibbles [rows][cols/2] + packed-dword
// e8m0 scales [cols/128][rows] (same scale layout as MX-fp8 round 2)
std::vector<torch::Tensor> mx4_convert(torch::Tensor a) {
  check_2d(a, "a");
  TORCH_CHECK(is_bf16(a), "mx4_convert needs bf16");
  long rows = a.size(0), cols = a.size(1);
  TORCH_CHECK(cols % 128 == 0, "cols must be a multiple of 128");
  auto opts = torch::TensorOptions().dtype(torch::kUInt8).device(a.device());
  auto out4 = torch::empty({rows, cols / 2}, opts);
  auto sc = torch::empty({cols / 128, rows},
                         torch::TensorOptions().dtype(torch::kInt32)
                             .device(a.device()));
  launch_mx4_convert(out4.data_ptr(), sc.data_ptr(), a.data_ptr(), rows,
                     cols, cur_stream());
  return {out4, sc};
}

// bf16 -> MX-fp6: packed e2m3 6-bit fields [rows][cols*3/4] + packed-dword
// e8m0 scales [cols/128][rows] (same scale layout as MX-fp8/fp4)
std::vector<torch::Tensor> mx6_convert(torch::Tensor a) {
  check_2d(a, "a");
  TORCH_CHECK(is_bf16(a), "mx6_convert needs bf16");
  long rows = a.size(0), cols = a.size(1);
  TORCH_CHECK(cols % 128 == 0, "cols must be a multiple of 128");
  auto opts = torch::TensorOptions().dtype(torch::kUInt8).device(a.device());
  auto out6 = torch::empty({rows, cols * 3 / 4}, opts);
  auto sc = torch::empty({cols / 128, rows},
                         torch::TensorOptions().dtype(torch::kInt32)
                             .device(a.device()));
  launch_mx6_convert(out6.data_ptr(), sc.data_ptr(), a.data_ptr(), rows,
                     cols, cur_stream());
  return {out6, sc};
}

// C(f32) += A6 @ Bt6^T with per-32-block e8m0 scales (MX-fp6 e2m3:
// fp8-class accuracy at the fp4 MFMA rate; opt-in)
void gemm_bt_mx6_acc(torch::Tensor C, torch::Tensor A6, torch::Tensor Asc,
                     torch::Tensor Bt6, torch::Tensor Bsc, bool beta0) {
  check_2d(C, "C");
  check_2d(A6, "A6");
  check_2d(Bt6, "Bt6");
  TORCH_CHECK(C.scalar_type() == torch::kFloat32);
  TORCH_CHECK(A6.scalar_type() == torch::kUInt8 &&
              Bt6.scalar_type() == torch::kUInt8);
  int M = (int)A6.size(0), K = (int)(A6.size(1) * 4 / 3),
      N = (int)Bt6.size(0);
  TORCH_CHECK(Bt6.size(1) * 4 / 3 == K && C.size(0) == M && C.size(1) == N);
  TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 128 == 0,
              "mx6 GEMM needs M,N % 256 == 0, K % 128 == 0");
  TORCH_CHECK(Asc.scalar_type() == torch::kInt32 &&
              Bsc.scalar_type() == torch::kInt32 &&
              Asc.size(0) == K / 128 && Asc.size(1) == M &&
              Bsc.size(0) == K / 128 && Bsc.size(1) == N,
              "packed scales must be [K/128][rows] int32");
  launch_gemm_bt_mx6_acc(C.data_ptr<float>(), A6.data_ptr(),
                         Asc.data_ptr(), Bt6.data_ptr(), Bsc.data_ptr(),
                         M, N, K, beta0 ? 1 : 0, cur_stream());
}

// C(f32) += A4 @ Bt4^T with per-32-block e8m0 scales (MX-fp4, ~2x the
// MX-fp8 rate at ~1-2 significant digits; opt-in)
void gemm_bt_mx4_acc(torch::Tensor C, torch::Tensor A4, torch::Tensor Asc,
                     torch::Tensor Bt4, torch::Tensor Bsc, bool beta0) {
  check_2d(C, "C");
  check_2d(A4, "A4");
  check_2d(Bt4, "Bt4");
  TORCH_CHECK(C.scalar_type() == torch::kFloat32);
  TORCH_CHECK(A4.scalar_type() == torch::kUInt8 &&
              Bt4.scalar_type() == torch::kUInt8);
  int M = (int)A4.size(0), K = (int)(A4.size(1) * 2), N = (int)Bt4.size(0);
  TORCH_CHECK(Bt4.size(1) * 2 == K && C.size(0) == M && C.size(1) == N);
  TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 128 == 0,
              "mx4 GEMM needs M,N % 256 == 0, K % 128 == 0");
  TORCH_CHECK(Asc.scalar_type() == torch::kInt32 &&
              Bsc.scalar_type() == torch::kInt32 &&
              Asc.size(0) == K / 128 && Asc.size(1) == M &&
              Bsc.size(0) == K / 128 && Bsc.size(1) == N,
              "packed scales must be [K/128][rows] int32");
  launch_gemm_bt_mx4_acc(C.data_ptr<float>(), A4.data_ptr(),
                         Asc.data_ptr(), Bt4.data_ptr(), Bsc.data_ptr(),
                         M, N, K, beta0 ? 1 : 0, cur_stream());
}

// C(f32) += A8 @ Bt8^T with per-32-block e8m0 scales
void gemm_bt_mx8_acc(torch::Tensor C, torch::Tensor A8, torch::Tensor Asc,
                     torch::Tensor Bt8, torch::Tensor Bsc, bool beta0) {
  check_2d(C, "C");
  check_2d(A8, "A8");
  check_2d(Bt8, "Bt8");
  TORCH_CHECK(C.scalar_type() == torch::kFloat32);
  TORCH_CHECK(A8.scalar_type() == torch::kUInt8 &&
              Bt8.scalar_type() == torch::kUInt8);
  int M = (int)A8.size(0), K = (int)A8.size(1), N = (int)Bt8.size(0);
  TORCH_CHECK(Bt8.size(1) == K && C.size(0) == M && C.size(1) == N);
  TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 64 == 0,
              "mx8 GEMM needs M,N % 256 == 0, K % 64 == 0");
  if (K % 128 == 0) {
    // packed-dword layout [K/128][rows] int32 (mx8_convert emits it for
    // cols % 128 == 0 inputs)
    TORCH_CHECK(Asc.scalar_type() == torch::kInt32 &&
                Bsc.scalar_type() == torch::kInt32,
                "K % 128 == 0 expects packed int32 scales from "
                "mx8_convert");
    TORCH_CHECK(Asc.size(0) == K / 128 && Asc.size(1) == M &&
                Bsc.size(0) == K / 128 && Bsc.size(1) == N,
                "packed scales must be [K/128][rows] int32");
  } else {
    TORCH_CHECK(Asc.numel() == (long)M * (K / 32) &&
                Bsc.numel() == (long)N * (K / 32));
    TORCH_CHECK(Asc.size(0) == K / 32 && Bsc.size(0) == K / 32,
                "scales must be block-major [K/32][rows]");
  }
  launch_gemm_bt_mx8_acc(C.data_ptr<float>(), A8.data_ptr(), Asc.data_ptr(),
                         Bt8.data_ptr(), Bsc.data_ptr(), M, N, K,
                         beta0 ? 1 : 0, cur_stream());
}

// out = (a OP b)^T in one pass (fused elementwise + transpose)
torch::Tensor ewise_transpose(int64_t op, torch::Tensor a, torch::Tensor b) {
  check_2d(a, "a");
  check_2d(b, "b");
  TORCH_CHECK(a.sizes() == b.sizes() && a.scalar_type() == b.scalar_type());
  auto out = torch::empty({a.size(1), a.size(0)}, a.options());
  if (is_bf16(a))
    launch_ewise_transpose_bf16((int)op, out.data_ptr(), a.data_ptr(),
                                b.data_ptr(), a.size(0), a.size(1),
                                cur_stream());
  else
    launch_ewise_transpose_f32((int)op, out.data_ptr<float>(),
                               a.data_ptr<float>(), b.data_ptr<float>(),
                               a.size(0), a.size(1), cur_stream());
  return out;
}

torch::Tensor transpose(torch::Tensor a) {
  check_2d(a, "a");
  auto out = torch::empty({a.size(1), a.size(0)}, a.options());
  if (is_bf16(a))
    launch_transpose_bf16(out.data_ptr(), a.data_ptr(), a.size(0), a.size(1),
                          cur_stream());
  else
    launch_transpose_f32(out.data_ptr<float>(), a.data_ptr<float>(),
                         a.size(0), a.size(1), cur_stream());
  return out;
}

// ---- native batch executor -------------------------------------------
// Executes a whole block-matmul task list without returning to Python
// between launches (the reference's DAGScheduler-task-loop equivalent for
// the per-block inner loop; SURVEY.md §2.2 "host-side DAG executor").
// Pays off in the fine-block regime (block-sparse plans with thousands of
// sub-millisecond GEMMs) where per-call Python dispatch is measurable.
//
// tasks: int64 [n][4] rows (acc_idx, a_idx, bt_idx, beta0).
void gemm_bt_batch(std::vector<torch::Tensor> accs,
                   std::vector<torch::Tensor> as,
                   std::vector<torch::Tensor> bts, torch::Tensor tasks) {
  TORCH_CHECK(tasks.dim() == 2 && tasks.size(1) == 4,
              "tasks must be [n][4]");
  TORCH_CHECK(tasks.scalar_type() == torch::kInt64);
  auto t = tasks.cpu().contiguous();
  const int64_t* p = t.data_ptr<int64_t>();
  int64_t n = t.size(0);
  hipStream_t stream = cur_stream();
  for (int64_t i = 0; i < n; ++i) {
    auto& C = accs.at((size_t)p[i * 4 + 0]);
    auto& A = as.at((size_t)p[i * 4 + 1]);
    auto& Bt = bts.at((size_t)p[i * 4 + 2]);
    bool beta0 = p[i * 4 + 3] != 0;
    int M = (int)A.size(0), K = (int)A.size(1), N = (int)Bt.size(0);
    TORCH_CHECK(C.size(0) == M && C.size(1) == N && Bt.size(1) == K);
    TORCH_CHECK(M % 128 == 0 && N % 128 == 0 && K % 64 == 0,
                "batch tasks must be tile-aligned");
    if (is_bf16(A)) {
      if (beta0)
        launch_gemm_bt_bf16_acc_b0(C.data_ptr<float>(), A.data_ptr(),
                                   Bt.data_ptr(), M, N, K, stream);
      else
        launch_gemm_bt_bf16_acc(C.data_ptr<float>(), A.data_ptr(),
                                Bt.data_ptr(), M, N, K, stream);
    } else {
      if (beta0) C.zero_();
      TORCH_CHECK(K % 32 == 0);
      launch_gemm_bt_f32_acc(C.data_ptr<float>(), A.data_ptr<float>(),
                             Bt.data_ptr<float>(), M, N, K, stream);
    }
  }
}

// batched transpose (B-panel preparation without Python round trips)
std::vector<torch::Tensor> transpose_batch(std::vector<torch::Tensor> ts) {
  std::vector<torch::Tensor> out;
  out.reserve(ts.size());
  for (auto& a : ts) out.push_back(transpose(a));
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "matrel_amd hand-written CDNA4 (gfx950) kernels";
  m.def("gemm_bt_acc", &gemm_bt_acc,
        "C(f32) += A @ Bt^T (bf16 MFMA / exact-f32 MFMA)");
  m.def("gemm_bt_acc_beta0", &gemm_bt_acc_beta0,
        "C(f32) = A @ Bt^T (C may be uninitialized)");
  m.def("gemm_bt_final",
        [](torch::Tensor C, torch::Tensor Co, torch::Tensor A,
           torch::Tensor Bt, bool beta0) -> bool {
          check_2d(C, "C");
          check_2d(Co, "Co");
          check_2d(A, "A");
          check_2d(Bt, "Bt");
          TORCH_CHECK(C.scalar_type() == torch::kFloat32 &&
                      Co.scalar_type() == torch::kBFloat16 &&
                      is_bf16(A) && is_bf16(Bt));
          int M = (int)A.size(0), K = (int)A.size(1), N = (int)Bt.size(0);
          TORCH_CHECK(Co.size(0) == M && Co.size(1) == N &&
                      C.size(0) == M && C.size(1) == N);
          TORCH_CHECK(M % 128 == 0 && N % 128 == 0 && K % 64 == 0);
          return launch_gemm_bt_bf16_final(
                     C.data_ptr<float>(), Co.data_ptr(), A.data_ptr(),
                     Bt.data_ptr(), M, N, K, beta0 ? 1 : 0,
                     cur_stream()) != 0;
        },
        "last-k fused: bf16 out straight from the accumulate epilogue; "
        "returns false when geometry unsupported (caller falls back)");
  m.def("set_f32_variant", [](int64_t v) { set_f32_variant((int)v); },
        "f32 GEMM variant (0 = 32x32x2 default, 1 = 16x16x4 256-tile)");
  m.def("set_mx8_variant", [](int64_t v) { set_mx8_variant((int)v); },
        "mx8 GEMM variant (0 auto, 2 counted 2-phase)");
  m.def("set_gemm_variant", [](int64_t v) { set_gemm_variant((int)v); },
        "0=auto 1=simple256 2=phased256 3=tile128 (A/B testing)");
  m.def("set_supertile", [](int64_t sm, int64_t sn) {
          set_supertile((int)sm, (int)sn);
        }, "XCD supertile cluster dims (0 disables)");
  m.def("ewise", &ewise, "elementwise a OP b");
  m.def("ewise_scalar", &ewise_scalar, "elementwise a OP s");
  m.def("mask_value", &mask_value, "value-predicate mask");
  m.def("mask_rows_cols", &mask_rows_cols, "index-predicate mask");
  m.def("outer", &outer, "outer cross-join");
  m.def("mask_isin", &mask_isin, "value semi-join mask");
  m.def("reduce2d", &reduce2d, "full/row/col reduction");
  m.def("trace", &trace, "diagonal sum");
  m.def("transpose", &transpose, "materialized transpose");
  m.def("ewise_transpose", &ewise_transpose,
        "(a OP b)^T fused single pass");
  m.def("gemm_bt_batch", &gemm_bt_batch,
        "execute a block-GEMM task list natively (no Python per launch)");
  m.def("transpose_batch", &transpose_batch, "batched transpose");
  m.def("spmm_csr_acc", &spmm_csr_acc, "C(f32) += CSR @ B (within-block sparse)");
  m.def("spgemm_csr_csr_acc", &spgemm_csr_csr_acc,
        "C(f32) += A_csr @ B_csr (sparse x sparse block product)");
  m.def("mx8_convert", &mx8_convert, "bf16 -> (e4m3, e8m0 block scales)");
  m.def("mx4_convert", &mx4_convert,
        "bf16 -> (e2m1 nibbles, e8m0 block scales)");
  m.def("gemm_bt_mx4_acc", &gemm_bt_mx4_acc,
        "C(f32) += A4 @ Bt4^T (MX-fp4 scaled MFMA, ~2x the fp8 rate)");
  m.def("mx6_convert", &mx6_convert,
        "bf16 -> (e2m3 6-bit fields, e8m0 block scales)");
  m.def("gemm_bt_mx6_acc", &gemm_bt_mx6_acc,
        "C(f32) += A6 @ Bt6^T (MX-fp6 scaled MFMA at the fp4 rate)");
  m.def("gemm_bt_mx8_acc", &gemm_bt_mx8_acc,
        "C(f32) += A8 @ Bt8^T (MX-fp8 scaled MFMA, ~2x bf16 rate)");
}
