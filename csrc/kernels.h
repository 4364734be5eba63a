// Shared declarations between the HIP kernel TUs and the torch extension TU.
//
// Design (see SURVEY.md §2.4/§2.5 for the reference per-tile op inventory this
// replaces): instead of one kernel launch per tile task, the hot ops are FUSED —
// one launch processes a whole list of tile-triples described by GemmDesc records,
// so a full trailing update / panel solve / back-transform sweep is a single
// kernel with >> 256 workgroups (MI355X: 256 CUs over 8 XCDs).
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

// op codes (match dlaf_amd.types.Op encoding used by the Python layer)
#define OP_N 0
#define OP_T 1
#define OP_C 2

// One fused-GEMM work item: C[c_off] += alpha * op(A[c_off]) * op(B[b_off]).
// Offsets are in ELEMENTS from the base pointers. A K-loop over `ktiles`
// operand tiles with the given element strides runs inside the kernel.
struct GemmDesc {
  int64_t c_off;
  int64_t a_off;
  int64_t b_off;
  int64_t ktiles;     // number of K tiles (>=1); total K = ktiles * K_param
  int64_t a_kstride;  // element stride between consecutive K tiles of A
  int64_t b_kstride;  // element stride between consecutive K tiles of B
};

extern "C" {

// ---- fused batched GEMM (MFMA f64 / f32) ----
// C (M x N, ldc), op(A) (M x K), op(B) (K x N); K is the per-tile K extent.
// `inplace` selects the wide-BN instantiation that makes X = X*op(B) safe in
// place (C block == A block); required whenever C aliases A.
void gemm_tiles_f64(const GemmDesc* descs, int ndesc, const double* A,
                    const double* B, double* C, int M, int N, int K, int lda,
                    int ldb, int ldc, int opA, int opB, double alpha,
                    double beta, hipStream_t stream, int inplace);
void gemm_tiles_f32(const GemmDesc* descs, int ndesc, const float* A,
                    const float* B, float* C, int M, int N, int K, int lda,
                    int ldb, int ldc, int opA, int opB, float alpha, float beta,
                    hipStream_t stream, int inplace);
// complex variants (interleaved re/im); opA/opB support OP_C (conjugate).
void gemm_tiles_c128(const GemmDesc* descs, int ndesc, const double* A,
                     const double* B, double* C, int M, int N, int K, int lda,
                     int ldb, int ldc, int opA, int opB, double alpha_re,
                     double alpha_im, double beta_re, double beta_im,
                     hipStream_t stream);
void gemm_tiles_c64(const GemmDesc* descs, int ndesc, const float* A,
                    const float* B, float* C, int M, int N, int K, int lda,
                    int ldb, int ldc, int opA, int opB, float alpha_re,
                    float alpha_im, float beta_re, float beta_im,
                    hipStream_t stream);

// ---- fused batched GEMM v2 (full-tile glds fast path; returns 1 if taken) ----
int gemm_tiles_v2_f64(const GemmDesc* descs, int ndesc, const double* A,
                      const double* B, double* C, int M, int N, int K, int lda,
                      int ldb, int ldc, int opA, int opB, double alpha,
                      double beta, hipStream_t stream);
int gemm_tiles_v2_f32(const GemmDesc* descs, int ndesc, const float* A,
                      const float* B, float* C, int M, int N, int K, int lda,
                      int ldb, int ldc, int opA, int opB, float alpha,
                      float beta, hipStream_t stream);
int gemm_tiles_v2_c128(const GemmDesc* descs, int ndesc, const double* A,
                       const double* B, double* C, int M, int N, int K,
                       int lda, int ldb, int ldc, int opA, int opB,
                       double alpha_re, double alpha_im, double beta_re,
                       double beta_im, hipStream_t stream);
int gemm_tiles_v2_c64(const GemmDesc* descs, int ndesc, const float* A,
                      const float* B, float* C, int M, int N, int K, int lda,
                      int ldb, int ldc, int opA, int opB, float alpha_re,
                      float alpha_im, float beta_re, float beta_im,
                      hipStream_t stream);

// ---- bt window-chain group apply (one launch per sweep group) ----
int bt_apply_group_f64(double* E, int64_t nE, int64_t npad, const double* V,
                       const double* VTt, int64_t base0, int b, int G, int R,
                       int nwin, hipStream_t stream);
int bt_apply_group_c128(double* E, int64_t nE, int64_t npad, const double* V,
                        const double* VTt, int64_t base0, int b, int G, int R,
                        int nwin, hipStream_t stream);

// ---- single-tile factorization building blocks ----
// Fused single-workgroup [factor +] invert of one diagonal block: factors the
// leading n x n (if do_factor) in place and writes its inverse into the
// BSZ x BSZ block Tout (BSZ = 128 real / 64 complex; identity-extended).
void potrf_invert_block_f64(double* A, int n, int ld, double* Tout,
                            int do_factor, hipStream_t stream);
void potrf_invert_block_f32(float* A, int n, int ld, float* Tout,
                            int do_factor, hipStream_t stream);
void potrf_invert_block_c128(double* A, int n, int ld, double* Tout,
                             int do_factor, hipStream_t stream);
void potrf_invert_block_c64(float* A, int n, int ld, float* Tout,
                            int do_factor, hipStream_t stream);

// In-place Cholesky (Lower) of the leading n x n (n <= 128) of a tile with row
// stride ld; single workgroup, LDS-resident.
void potrf_block128_f64(double* A, int n, int ld, hipStream_t stream);
void potrf_block128_f32(float* A, int n, int ld, hipStream_t stream);
void potrf_block128_c128(double* A, int n, int ld, hipStream_t stream);
void potrf_block128_c64(float* A, int n, int ld, hipStream_t stream);

// Column-parallel inverse of a lower-triangular n x n block (n <= 128 per
// workgroup column count; arbitrary n supported): T = L^-1, T may not alias L.
void trtri_lower_f64(const double* L, double* T, int n, int ldl, int ldt,
                     int unit_diag, hipStream_t stream);
void trtri_lower_f32(const float* L, float* T, int n, int ldl, int ldt,
                     int unit_diag, hipStream_t stream);
void trtri_lower_c128(const double* L, double* T, int n, int ldl, int ldt,
                      int unit_diag, hipStream_t stream);
void trtri_lower_c64(const float* L, float* T, int n, int ldl, int ldt,
                     int unit_diag, hipStream_t stream);

}  // extern "C"
