// Torch extension bindings for the dlaf_amd CDNA4 kernels.
//
// All entry points take CUDA (ROCm) tensors, run on the CURRENT torch stream
// (so the Python runtime's stream/event scheduling applies), and are
// asynchronous. The per-tile Cholesky (`potrf_tile`) is host-orchestrated here
// in C++: potrf_block (single-WG LDS kernel) + column-parallel block inverse +
// fused-GEMM panel/trailing updates, with all GemmDescs staged to the device in
// ONE H2D copy per call.
#include <torch/extension.h>
#include <c10/cuda/CUDAStream.h>
#include <c10/cuda/CUDACachingAllocator.h>
#include <hip/hip_runtime.h>

#include <vector>

#include "kernels.h"

// csrc/chase_gpu.hip
extern "C" {
void chase_gpu_f64(double*, int64_t, int64_t, int64_t, double*, const int64_t*,
                   int32_t*, int32_t*, hipStream_t);
void chase_gpu_f32(float*, int64_t, int64_t, int64_t, float*, const int64_t*,
                   int32_t*, int32_t*, hipStream_t);
void chase_gpu_c128(double*, int64_t, int64_t, int64_t, double*, const int64_t*,
                    int32_t*, int32_t*, hipStream_t);
void chase_gpu_c64(float*, int64_t, int64_t, int64_t, float*, const int64_t*,
                   int32_t*, int32_t*, hipStream_t);
}

// csrc/rocblas_batch.cpp
void lib_gemm_batched(torch::Tensor dt, torch::Tensor ptrC, torch::Tensor ptrA,
                      torch::Tensor ptrB, int64_t M, int64_t N, int64_t K,
                      int64_t lda, int64_t ldb, int64_t ldc, int64_t opA,
                      int64_t opB, double alpha_re, double alpha_im,
                      double beta_re, double beta_im);

namespace {

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e));    \
  } while (0)

hipStream_t cur_stream() {
  return (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
}

int potrf_bsz(at::ScalarType t) {
  (void)t;
  return 64;  // see potrf_invert_block launchers: 64 keeps the kernel co-residable
}

void check_gemm_args(const torch::Tensor& desc, const torch::Tensor& A,
                     const torch::Tensor& B, const torch::Tensor& C) {
  TORCH_CHECK(C.is_cuda() && A.is_cuda() && B.is_cuda(), "tensors must be on GPU");
  TORCH_CHECK(desc.is_cuda() && desc.scalar_type() == at::kLong && desc.is_contiguous(),
              "desc must be contiguous int64 on GPU");
  TORCH_CHECK(desc.dim() == 2 && desc.size(1) == 6, "desc must be [n, 6]");
  TORCH_CHECK(A.scalar_type() == C.scalar_type() && B.scalar_type() == C.scalar_type(),
              "dtype mismatch");
}

// C[desc] = alpha * op(A) op(B) + beta * C ; desc rows are
// (c_off, a_off, b_off, ktiles, a_kstride, b_kstride) in element units.
void batch_gemm(torch::Tensor C, torch::Tensor A, torch::Tensor B,
                torch::Tensor desc, int64_t M, int64_t N, int64_t K,
                int64_t lda, int64_t ldb, int64_t ldc, int64_t opA, int64_t opB,
                double alpha_re, double alpha_im, double beta_re,
                double beta_im, bool inplace) {
  check_gemm_args(desc, A, B, C);
  const int nd = (int)desc.size(0);
  if (nd == 0) return;
  auto descs = reinterpret_cast<const GemmDesc*>(desc.data_ptr<int64_t>());
  auto s = cur_stream();
  switch (C.scalar_type()) {
    case at::kDouble:
      gemm_tiles_f64(descs, nd, A.data_ptr<double>(), B.data_ptr<double>(),
                     C.data_ptr<double>(), M, N, K, lda, ldb, ldc, opA, opB,
                     alpha_re, beta_re, s, inplace);
      break;
    case at::kFloat:
      gemm_tiles_f32(descs, nd, A.data_ptr<float>(), B.data_ptr<float>(),
                     C.data_ptr<float>(), M, N, K, lda, ldb, ldc, opA, opB,
                     (float)alpha_re, (float)beta_re, s, inplace);
      break;
    case at::kComplexDouble:
      gemm_tiles_c128(descs, nd, (const double*)A.data_ptr(),
                      (const double*)B.data_ptr(), (double*)C.data_ptr(), M, N,
                      K, lda, ldb, ldc, opA, opB, alpha_re, alpha_im, beta_re,
                      beta_im, s);
      break;
    case at::kComplexFloat:
      gemm_tiles_c64(descs, nd, (const float*)A.data_ptr(),
                     (const float*)B.data_ptr(), (float*)C.data_ptr(), M, N, K,
                     lda, ldb, ldc, opA, opB, (float)alpha_re, (float)alpha_im,
                     (float)beta_re, (float)beta_im, s);
      break;
    default:
      TORCH_CHECK(false, "unsupported dtype");
  }
  HIP_CHECK(hipGetLastError());
}

void potrf_block(torch::Tensor A, int64_t n, int64_t ld) {
  TORCH_CHECK(A.is_cuda());
  auto s = cur_stream();
  switch (A.scalar_type()) {
    case at::kDouble:
      TORCH_CHECK(n <= 128);
      potrf_block128_f64(A.data_ptr<double>(), n, ld, s);
      break;
    case at::kFloat:
      TORCH_CHECK(n <= 128);
      potrf_block128_f32(A.data_ptr<float>(), n, ld, s);
      break;
    case at::kComplexDouble:
      TORCH_CHECK(n <= 64);
      potrf_block128_c128((double*)A.data_ptr(), n, ld, s);
      break;
    case at::kComplexFloat:
      TORCH_CHECK(n <= 64);
      potrf_block128_c64((float*)A.data_ptr(), n, ld, s);
      break;
    default:
      TORCH_CHECK(false, "unsupported dtype");
  }
  HIP_CHECK(hipGetLastError());
}

// T = tril(L)^-1 for an n x n view; T must not alias L.
void trtri_lower(torch::Tensor L, torch::Tensor T, int64_t n, int64_t ldl,
                 int64_t ldt, bool unit_diag) {
  TORCH_CHECK(L.is_cuda() && T.is_cuda());
  auto s = cur_stream();
  switch (L.scalar_type()) {
    case at::kDouble:
      trtri_lower_f64(L.data_ptr<double>(), T.data_ptr<double>(), n, ldl, ldt,
                      unit_diag, s);
      break;
    case at::kFloat:
      trtri_lower_f32(L.data_ptr<float>(), T.data_ptr<float>(), n, ldl, ldt,
                      unit_diag, s);
      break;
    case at::kComplexDouble:
      trtri_lower_c128((const double*)L.data_ptr(), (double*)T.data_ptr(), n,
                       ldl, ldt, unit_diag, s);
      break;
    case at::kComplexFloat:
      trtri_lower_c64((const float*)L.data_ptr(), (float*)T.data_ptr(), n, ldl,
                      ldt, unit_diag, s);
      break;
    default:
      TORCH_CHECK(false, "unsupported dtype");
  }
  HIP_CHECK(hipGetLastError());
}

// In-place Cholesky (Lower) of the leading n x n of a tile with row stride ld.
// Also fills `dinv` ([nblocks, bsz, bsz] contiguous) with the inverses of the
// bsz x bsz diagonal blocks of the factor — the panel TRSM then becomes GEMMs.
// `ddesc` is the prebuilt device descriptor table ([2*nblocks, 6] int64,
// cached by the Python layer per (n, ld, dtype)): rows 2d   = panel desc
// (A21 offsets), rows 2d+1 = trailing desc, exactly as _potrf_descs builds.
void potrf_tile(torch::Tensor A, int64_t n, int64_t ld, torch::Tensor dinv,
                torch::Tensor ddesc) {
  TORCH_CHECK(A.is_cuda() && dinv.is_cuda() && dinv.is_contiguous());
  const int bsz = potrf_bsz(A.scalar_type());
  const int nblocks = (int)((n + bsz - 1) / bsz);
  TORCH_CHECK(dinv.numel() >= (int64_t)nblocks * bsz * bsz, "dinv too small");
  TORCH_CHECK(ddesc.is_cuda() && ddesc.scalar_type() == at::kLong &&
              ddesc.is_contiguous() && ddesc.numel() >= nblocks * 12,
              "bad ddesc");
  auto s = cur_stream();
  auto descs = reinterpret_cast<const GemmDesc*>(ddesc.data_ptr<int64_t>());

  for (int d = 0; d < nblocks; ++d) {
    const int64_t c0 = (int64_t)d * bsz;
    const int bs = (int)std::min<int64_t>(bsz, n - c0);
    // 1+2) fused: factor diagonal block and write its inverse -> dinv[d]
    {
      const int64_t off = c0 * ld + c0;
      const int64_t doff = (int64_t)d * bsz * bsz;
      switch (A.scalar_type()) {
        case at::kDouble:
          potrf_invert_block_f64(A.data_ptr<double>() + off, bs, ld,
                                 dinv.data_ptr<double>() + doff, 1, s);
          break;
        case at::kFloat:
          potrf_invert_block_f32(A.data_ptr<float>() + off, bs, ld,
                                 dinv.data_ptr<float>() + doff, 1, s);
          break;
        case at::kComplexDouble:
          potrf_invert_block_c128((double*)A.data_ptr() + 2 * off, bs, ld,
                                  (double*)dinv.data_ptr() + 2 * doff, 1, s);
          break;
        case at::kComplexFloat:
          potrf_invert_block_c64((float*)A.data_ptr() + 2 * off, bs, ld,
                                 (float*)dinv.data_ptr() + 2 * doff, 1, s);
          break;
        default:
          TORCH_CHECK(false);
      }
    }
    const int64_t rows_below = n - c0 - bs;
    if (rows_below <= 0) continue;
    const GemmDesc* dp = descs + 2 * d;
    const int64_t doff = (int64_t)d * bsz * bsz;
    // 3) panel: X = A21 * dinv^H  (in place)
    // 4) trailing: A22 -= X X^H
    switch (A.scalar_type()) {
      case at::kDouble: {
        auto Ap = A.data_ptr<double>();
        auto Dp = dinv.data_ptr<double>() + doff;
        gemm_tiles_f64(dp, 1, Ap, Dp, Ap, rows_below, bs, bs, ld, bsz, ld, OP_N,
                       OP_T, 1.0, 0.0, s, 1);
        gemm_tiles_f64(dp + 1, 1, Ap, Ap, Ap, rows_below, rows_below, bs, ld,
                       ld, ld, OP_N, OP_T, -1.0, 1.0, s, 0);
        break;
      }
      case at::kFloat: {
        auto Ap = A.data_ptr<float>();
        auto Dp = dinv.data_ptr<float>() + doff;
        gemm_tiles_f32(dp, 1, Ap, Dp, Ap, rows_below, bs, bs, ld, bsz, ld, OP_N,
                       OP_T, 1.0f, 0.0f, s, 1);
        gemm_tiles_f32(dp + 1, 1, Ap, Ap, Ap, rows_below, rows_below, bs, ld,
                       ld, ld, OP_N, OP_T, -1.0f, 1.0f, s, 0);
        break;
      }
      case at::kComplexDouble: {
        auto Ap = (double*)A.data_ptr();
        auto Dp = (double*)dinv.data_ptr() + 2 * doff;
        gemm_tiles_c128(dp, 1, Ap, Dp, Ap, rows_below, bs, bs, ld, bsz, ld,
                        OP_N, OP_C, 1.0, 0.0, 0.0, 0.0, s);
        gemm_tiles_c128(dp + 1, 1, Ap, Ap, Ap, rows_below, rows_below, bs, ld,
                        ld, ld, OP_N, OP_C, -1.0, 0.0, 1.0, 0.0, s);
        break;
      }
      case at::kComplexFloat: {
        auto Ap = (float*)A.data_ptr();
        auto Dp = (float*)dinv.data_ptr() + 2 * doff;
        gemm_tiles_c64(dp, 1, Ap, Dp, Ap, rows_below, bs, bs, ld, bsz, ld, OP_N,
                       OP_C, 1.0f, 0.0f, 0.0f, 0.0f, s);
        gemm_tiles_c64(dp + 1, 1, Ap, Ap, Ap, rows_below, rows_below, bs, ld,
                       ld, ld, OP_N, OP_C, -1.0f, 0.0f, 1.0f, 0.0f, s);
        break;
      }
      default:
        TORCH_CHECK(false);
    }
  }
  HIP_CHECK(hipGetLastError());
}

// Factor (optional) + invert ONE bsz-block of a tile view; Tout = dinv[d].
void factor_invert_block(torch::Tensor A, int64_t n, int64_t ld,
                         torch::Tensor Tout, bool do_factor) {
  TORCH_CHECK(A.is_cuda() && Tout.is_cuda());
  auto s = cur_stream();
  switch (A.scalar_type()) {
    case at::kDouble:
      TORCH_CHECK(n <= 64);
      potrf_invert_block_f64(A.data_ptr<double>(), n, ld,
                             Tout.data_ptr<double>(), do_factor, s);
      break;
    case at::kFloat:
      TORCH_CHECK(n <= 64);
      potrf_invert_block_f32(A.data_ptr<float>(), n, ld,
                             Tout.data_ptr<float>(), do_factor, s);
      break;
    case at::kComplexDouble:
      TORCH_CHECK(n <= 64);
      potrf_invert_block_c128((double*)A.data_ptr(), n, ld,
                              (double*)Tout.data_ptr(), do_factor, s);
      break;
    case at::kComplexFloat:
      TORCH_CHECK(n <= 64);
      potrf_invert_block_c64((float*)A.data_ptr(), n, ld,
                             (float*)Tout.data_ptr(), do_factor, s);
      break;
    default:
      TORCH_CHECK(false, "unsupported dtype");
  }
  HIP_CHECK(hipGetLastError());
}

}  // namespace

void band_chase(torch::Tensor band, int64_t b, torch::Tensor vstore,
                torch::Tensor offsets, int64_t nthreads);
int64_t dc_deflate_scan(torch::Tensor d, torch::Tensor z, double rho,
                        double tol, torch::Tensor deflated, torch::Tensor rots);

extern "C" {
void secular_roots_f64(const double*, const double*, int, double, long long*,
                       double*, hipStream_t);
int panel_qr_f64(double*, long, int, long, double*, double*, double*, hipStream_t);
int panel_qr_f32(float*, long, int, long, float*, float*, float*, hipStream_t);
int panel_qr_c128(double*, long, int, long, double*, double*, double*, hipStream_t);
int panel_qr_c64(float*, long, int, long, float*, float*, float*, hipStream_t);
}

// Cooperative whole-panel QR (see csrc/panel_qr.hip). P: 2D device view with
// unit column stride; taus: [min(m,nb)]; norms/wraw: zeroed workspaces.
void panel_qr(torch::Tensor P, torch::Tensor taus, torch::Tensor norms,
              torch::Tensor wraw) {
  TORCH_CHECK(P.is_cuda() && P.dim() == 2 && P.stride(1) == 1);
  long m = P.size(0);
  int nb = (int)P.size(1);
  long ldp = P.stride(0);
  auto stream = at::cuda::getCurrentHIPStream().stream();
  int rc = -1;
  switch (P.scalar_type()) {
    case torch::kFloat64:
      rc = panel_qr_f64((double*)P.data_ptr(), m, nb, ldp,
                        (double*)taus.data_ptr(), (double*)norms.data_ptr(),
                        (double*)wraw.data_ptr(), stream);
      break;
    case torch::kFloat32:
      rc = panel_qr_f32((float*)P.data_ptr(), m, nb, ldp,
                        (float*)taus.data_ptr(), (float*)norms.data_ptr(),
                        (float*)wraw.data_ptr(), stream);
      break;
    case torch::kComplexDouble:
      rc = panel_qr_c128((double*)P.data_ptr(), m, nb, ldp,
                         (double*)taus.data_ptr(), (double*)norms.data_ptr(),
                         (double*)wraw.data_ptr(), stream);
      break;
    case torch::kComplexFloat:
      rc = panel_qr_c64((float*)P.data_ptr(), m, nb, ldp,
                        (float*)taus.data_ptr(), (float*)norms.data_ptr(),
                        (float*)wraw.data_ptr(), stream);
      break;
    default:
      TORCH_CHECK(false, "unsupported dtype");
  }
  TORCH_CHECK(rc == 0, "panel_qr cooperative launch failed, hipError ", rc);
  HIP_CHECK(hipGetLastError());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("panel_qr", &panel_qr, "cooperative whole-panel QR (one launch)");
  m.def("secular_roots", [](torch::Tensor d, torch::Tensor z2, double rho,
                            torch::Tensor sidx, torch::Tensor mu) {
    TORCH_CHECK(d.is_cuda() && d.scalar_type() == torch::kFloat64);
    auto stream = at::cuda::getCurrentHIPStream().stream();
    secular_roots_f64(d.data_ptr<double>(), z2.data_ptr<double>(),
                      (int)d.size(0), rho, (long long*)sidx.data_ptr<int64_t>(),
                      mu.data_ptr<double>(), stream);
    HIP_CHECK(hipGetLastError());
  }, "D&C secular-equation roots, one thread per root");
  m.def("dc_deflate_scan", &dc_deflate_scan,
        "sequential deflation scan for the D&C merge");
  m.def("band_chase", &band_chase,
        "CPU bulge chasing band->tridiag with reflector recording",
        py::arg("band"), py::arg("b"), py::arg("vstore"), py::arg("offsets"),
        py::arg("nthreads") = 0);
  m.def("bt_apply_group", [](torch::Tensor E, torch::Tensor V,
                             torch::Tensor VTt, int64_t base0, int64_t b,
                             int64_t G, int64_t R, int64_t nwin) -> bool {
    TORCH_CHECK(E.is_cuda() && V.is_cuda() && VTt.is_cuda());
    TORCH_CHECK(E.is_contiguous() && V.is_contiguous() && VTt.is_contiguous());
    auto s = cur_stream();
    const int64_t npad = E.size(0), nE = E.size(1);
    int ok = 0;
    switch (E.scalar_type()) {
      case at::kDouble:
        ok = bt_apply_group_f64(E.data_ptr<double>(), nE, npad,
                                V.data_ptr<double>(), VTt.data_ptr<double>(),
                                base0, (int)b, (int)G, (int)R, (int)nwin, s);
        break;
      case at::kComplexDouble:
        ok = bt_apply_group_c128((double*)E.data_ptr(), nE, npad,
                                 (const double*)V.data_ptr(),
                                 (const double*)VTt.data_ptr(), base0, (int)b,
                                 (int)G, (int)R, (int)nwin, s);
        break;
      default:
        return false;
    }
    HIP_CHECK(hipGetLastError());
    return ok != 0;
  }, "whole-group bt window-chain apply");
  m.def("batch_gemm", &batch_gemm,
        "fused batched tile GEMM: C[d] = alpha*op(A[d])op(B[d]) + beta*C[d]");
  m.def("potrf_block", &potrf_block, "single-workgroup Cholesky block factor");
  m.def("factor_invert_block", &factor_invert_block,
        "fused single-workgroup [factor+]invert of a diagonal block");
  m.def("trtri_lower", &trtri_lower, "lower-triangular block inverse");
  m.def("band_chase_gpu", [](torch::Tensor band, int64_t b, torch::Tensor vstore,
                             torch::Tensor offsets, torch::Tensor done,
                             torch::Tensor abortf) {
    TORCH_CHECK(band.is_cuda() && vstore.is_cuda() && offsets.is_cuda() &&
                done.is_cuda() && abortf.is_cuda(), "device tensors required");
    TORCH_CHECK(b <= 64, "band_chase_gpu handles b <= 64");
    TORCH_CHECK(done.scalar_type() == at::kInt && abortf.scalar_type() == at::kInt);
    const int64_t size = band.size(0), ld = band.size(1);
    auto s = (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
    switch (band.scalar_type()) {
      case at::kDouble:
        chase_gpu_f64(band.data_ptr<double>(), ld, size, b,
                      vstore.data_ptr<double>(), offsets.data_ptr<int64_t>(),
                      done.data_ptr<int32_t>(), abortf.data_ptr<int32_t>(), s);
        break;
      case at::kFloat:
        chase_gpu_f32(band.data_ptr<float>(), ld, size, b,
                      vstore.data_ptr<float>(), offsets.data_ptr<int64_t>(),
                      done.data_ptr<int32_t>(), abortf.data_ptr<int32_t>(), s);
        break;
      case at::kComplexDouble:
        chase_gpu_c128((double*)band.data_ptr(), ld, size, b,
                       (double*)vstore.data_ptr(), offsets.data_ptr<int64_t>(),
                       done.data_ptr<int32_t>(), abortf.data_ptr<int32_t>(), s);
        break;
      case at::kComplexFloat:
        chase_gpu_c64((float*)band.data_ptr(), ld, size, b,
                      (float*)vstore.data_ptr(), offsets.data_ptr<int64_t>(),
                      done.data_ptr<int32_t>(), abortf.data_ptr<int32_t>(), s);
        break;
      default:
        TORCH_CHECK(false, "unsupported dtype");
    }
  }, "GPU bulge chase (persistent wavefront workgroups)");
  m.def("lib_gemm_batched", &lib_gemm_batched,
        "rocBLAS pointer-array batched GEMM (uniform tile shape)");
  m.def("potrf_tile", &potrf_tile,
        "in-place tile Cholesky + diagonal-block inverses");
  m.attr("POTRF_BSZ_REAL") = 64;
  m.attr("POTRF_BSZ_CPLX") = 64;
}
