// rocBLAS pointer-array batched GEMM for uniform-shape tile batches.
//
// The fused descriptor kernel (gemm_tiles.hip) covers arbitrary mixed-k
// batches in one launch, but rocBLAS's batched DGEMM runs the plain
// nb x nb x nb tile shape ~30% faster (57.8 vs 44.1 TF at 1024 tiles,
// profiles/microbench_r1.log). Algorithm phases whose batch is uniform
// (Cholesky trailing update, panel applies: one K-product per C tile)
// route here; mixed-ktiles phases keep the fused kernel.
//
// Library use is plain batched GEMM only — every fused/specialised op stays
// in the hand-written CDNA4 kernels. Reference counterpart: the per-tile
// cublas gemm calls of factorization/cholesky/impl.h:150-210.
//
// Row-major convention: rocBLAS is column-major, so we compute
// C_rm = op(A_rm) op(B_rm) as gemm(op(B), op(A), N, M, K, B, A, C) — same
// op flags, operands and dims swapped (buffer of a row-major matrix is the
// column-major buffer of its transpose).
//
// Pointer arrays are device int64 tensors computed in Python as
// base_data_ptr + element_offset * itemsize (cached per factorization shape
// next to the descriptor tables, so the hot loop uploads nothing).

#include <torch/extension.h>
#include <c10/cuda/CUDAStream.h>
#include <hip/hip_runtime.h>
#include <rocblas/rocblas.h>

namespace {

rocblas_handle handle_ = nullptr;

rocblas_handle get_handle() {
  if (!handle_) {
    TORCH_CHECK(rocblas_create_handle(&handle_) == rocblas_status_success,
                "rocblas_create_handle failed");
  }
  return handle_;
}

rocblas_operation to_op(int64_t o) {
  switch (o) {
    case 0: return rocblas_operation_none;
    case 1: return rocblas_operation_transpose;
    default: return rocblas_operation_conjugate_transpose;
  }
}

#define RB_CHECK(x) TORCH_CHECK((x) == rocblas_status_success, "rocblas: ", #x)

}  // namespace

// ptrC/ptrA/ptrB: device int64 tensors holding the DEVICE ADDRESSES of each
// batch entry's tile base. dt tensor is only consulted for the scalar type.
void lib_gemm_batched(torch::Tensor dt, torch::Tensor ptrC, torch::Tensor ptrA,
                      torch::Tensor ptrB, int64_t M, int64_t N, int64_t K,
                      int64_t lda, int64_t ldb, int64_t ldc, int64_t opA,
                      int64_t opB, double alpha_re, double alpha_im,
                      double beta_re, double beta_im) {
  const int64_t count = ptrC.numel();
  if (count == 0) return;
  TORCH_CHECK(ptrC.is_cuda() && ptrA.is_cuda() && ptrB.is_cuda(),
              "pointer arrays must be on device");
  TORCH_CHECK(ptrA.numel() == count && ptrB.numel() == count, "batch mismatch");
  auto h = get_handle();
  RB_CHECK(rocblas_set_stream(
      h, (hipStream_t)at::cuda::getCurrentCUDAStream().stream()));
  // row-major: swap operands and M/N, same op flags
  auto ta = to_op(opB), tb = to_op(opA);
  auto AA = reinterpret_cast<const void* const*>(ptrB.data_ptr<int64_t>());
  auto BB = reinterpret_cast<const void* const*>(ptrA.data_ptr<int64_t>());
  auto CC = reinterpret_cast<void* const*>(ptrC.data_ptr<int64_t>());
  const int m = (int)N, n = (int)M, k = (int)K;
  const int ld_a = (int)ldb, ld_b = (int)lda, ld_c = (int)ldc;
  switch (dt.scalar_type()) {
    case at::kDouble: {
      const double al = alpha_re, be = beta_re;
      RB_CHECK(rocblas_dgemm_batched(
          h, ta, tb, m, n, k, &al, (const double* const*)AA, ld_a,
          (const double* const*)BB, ld_b, &be, (double* const*)CC, ld_c,
          (int)count));
      break;
    }
    case at::kFloat: {
      const float al = (float)alpha_re, be = (float)beta_re;
      RB_CHECK(rocblas_sgemm_batched(
          h, ta, tb, m, n, k, &al, (const float* const*)AA, ld_a,
          (const float* const*)BB, ld_b, &be, (float* const*)CC, ld_c,
          (int)count));
      break;
    }
    case at::kComplexDouble: {
      const rocblas_double_complex al{alpha_re, alpha_im}, be{beta_re, beta_im};
      RB_CHECK(rocblas_zgemm_batched(
          h, ta, tb, m, n, k, &al, (const rocblas_double_complex* const*)AA,
          ld_a, (const rocblas_double_complex* const*)BB, ld_b, &be,
          (rocblas_double_complex* const*)CC, ld_c, (int)count));
      break;
    }
    case at::kComplexFloat: {
      const rocblas_float_complex al{(float)alpha_re, (float)alpha_im},
          be{(float)beta_re, (float)beta_im};
      RB_CHECK(rocblas_cgemm_batched(
          h, ta, tb, m, n, k, &al, (const rocblas_float_complex* const*)AA,
          ld_a, (const rocblas_float_complex* const*)BB, ld_b, &be,
          (rocblas_float_complex* const*)CC, ld_c, (int)count));
      break;
    }
    default:
      TORCH_CHECK(false, "unsupported dtype for lib_gemm_batched");
  }
}
