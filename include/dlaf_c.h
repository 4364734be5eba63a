/* C API of dlaf_amd — counterpart of the reference's include/dlaf_c/
 * (grid.h, desc.h, init.h, factorization/cholesky.h, inverse/cholesky.h,
 * eigensolver/eigensolver.h, eigensolver/gen_eigensolver.h).
 *
 * The library (libdlaf_c.so) embeds a Python interpreter and forwards to
 * the dlaf_amd package, which runs the native HIP/CDNA4 algorithms; the
 * caller's buffers are wrapped without copies (column-major, leading
 * dimension ld, ScaLAPACK convention).
 *
 * Grids: nprow x npcol contexts. For multi-process grids launch one
 * process per rank with the torchrun-style rendezvous environment set
 * (RANK, WORLD_SIZE, MASTER_ADDR, MASTER_PORT); dlaf_create_grid then
 * initializes torch.distributed inside the embedded runtime (RCCL when a
 * GPU is visible, gloo otherwise) and nprow*npcol must equal WORLD_SIZE.
 * The reference builds grids from an MPI_Comm / BLACS context
 * (src/c_api/grid.cpp); this framework's rank rendezvous is the launcher
 * environment instead of MPI — there is no MPI/BLACS interop in the
 * MI355X-native design. Buffers are the caller's rank-local ScaLAPACK
 * block-cyclic panels, exactly as in the reference shims.
 */
#ifndef DLAF_C_H
#define DLAF_C_H

#ifdef __cplusplus
extern "C" {
#endif

struct DLAF_descriptor {
  int m, n, mb, nb;
  int isrc, jsrc;
  int i, j;
  int ld;
};

typedef struct { float re, im; } dlaf_complex_c;
typedef struct { double re, im; } dlaf_complex_z;

/* init / teardown of the embedded runtime (reference: dlaf_c/init.h) */
int dlaf_initialize(int argc, const char* const* argv);
void dlaf_finalize(void);

/* grid management (reference: dlaf_c/grid.h); see header comment for
 * multi-process grids */
int dlaf_create_grid(int nprow, int npcol, char order);
void dlaf_free_grid(int ctx);

/* Cholesky factorization, lower triangle in place (dlaf_c/factorization/cholesky.h) */
int dlaf_cholesky_factorization_s(int ctx, char uplo, float* a, struct DLAF_descriptor desc);
int dlaf_cholesky_factorization_d(int ctx, char uplo, double* a, struct DLAF_descriptor desc);
int dlaf_cholesky_factorization_c(int ctx, char uplo, dlaf_complex_c* a, struct DLAF_descriptor desc);
int dlaf_cholesky_factorization_z(int ctx, char uplo, dlaf_complex_z* a, struct DLAF_descriptor desc);

/* A^-1 from the Cholesky factor (dlaf_c/inverse/cholesky.h) */
int dlaf_inverse_from_cholesky_factor_s(int ctx, char uplo, float* a, struct DLAF_descriptor desc);
int dlaf_inverse_from_cholesky_factor_d(int ctx, char uplo, double* a, struct DLAF_descriptor desc);
int dlaf_inverse_from_cholesky_factor_c(int ctx, char uplo, dlaf_complex_c* a, struct DLAF_descriptor desc);
int dlaf_inverse_from_cholesky_factor_z(int ctx, char uplo, dlaf_complex_z* a, struct DLAF_descriptor desc);

/* symmetric / Hermitian eigensolver (dlaf_c/eigensolver/eigensolver.h):
 * eigenvalues into w (ascending), eigenvectors into z. */
int dlaf_symmetric_eigensolver_s(int ctx, char uplo, float* a, struct DLAF_descriptor desca,
                                 float* w, float* z, struct DLAF_descriptor descz);
int dlaf_symmetric_eigensolver_d(int ctx, char uplo, double* a, struct DLAF_descriptor desca,
                                 double* w, double* z, struct DLAF_descriptor descz);
int dlaf_hermitian_eigensolver_c(int ctx, char uplo, dlaf_complex_c* a, struct DLAF_descriptor desca,
                                 float* w, dlaf_complex_c* z, struct DLAF_descriptor descz);
int dlaf_hermitian_eigensolver_z(int ctx, char uplo, dlaf_complex_z* a, struct DLAF_descriptor desca,
                                 double* w, dlaf_complex_z* z, struct DLAF_descriptor descz);

/* partial spectrum [il, iu): 0-based half-open like the reference's
 * eigenvalues_index_begin/end */
int dlaf_symmetric_eigensolver_partial_spectrum_d(
    int ctx, char uplo, double* a, struct DLAF_descriptor desca,
    double* w, double* z, struct DLAF_descriptor descz, long il, long iu);
int dlaf_hermitian_eigensolver_partial_spectrum_z(
    int ctx, char uplo, dlaf_complex_z* a, struct DLAF_descriptor desca,
    double* w, dlaf_complex_z* z, struct DLAF_descriptor descz, long il, long iu);

/* generalized eigensolver A x = lambda B x (dlaf_c/eigensolver/gen_eigensolver.h);
 * the _factorized variants take B already Cholesky-factorized. */
int dlaf_symmetric_generalized_eigensolver_d(
    int ctx, char uplo, double* a, struct DLAF_descriptor desca,
    double* b, struct DLAF_descriptor descb,
    double* w, double* z, struct DLAF_descriptor descz);
int dlaf_symmetric_generalized_eigensolver_factorized_d(
    int ctx, char uplo, double* a, struct DLAF_descriptor desca,
    double* b, struct DLAF_descriptor descb,
    double* w, double* z, struct DLAF_descriptor descz);
int dlaf_hermitian_generalized_eigensolver_z(
    int ctx, char uplo, dlaf_complex_z* a, struct DLAF_descriptor desca,
    dlaf_complex_z* b, struct DLAF_descriptor descb,
    double* w, dlaf_complex_z* z, struct DLAF_descriptor descz);
int dlaf_hermitian_generalized_eigensolver_factorized_z(
    int ctx, char uplo, dlaf_complex_z* a, struct DLAF_descriptor desca,
    dlaf_complex_z* b, struct DLAF_descriptor descb,
    double* w, dlaf_complex_z* z, struct DLAF_descriptor descz);

/* ScaLAPACK-style shims (dlaf_pdpotrf etc.): desca is the 9-int ScaLAPACK
 * descriptor (DTYPE_, CTXT_, M_, N_, MB_, NB_, RSRC_, CSRC_, LLD_). */
void dlaf_pdpotrf(char uplo, int n, double* a, int ia, int ja, const int desca[9], int* info);
void dlaf_pspotrf(char uplo, int n, float* a, int ia, int ja, const int desca[9], int* info);
void dlaf_pzpotrf(char uplo, int n, dlaf_complex_z* a, int ia, int ja, const int desca[9], int* info);
void dlaf_pcpotrf(char uplo, int n, dlaf_complex_c* a, int ia, int ja, const int desca[9], int* info);
void dlaf_pdpotri(char uplo, int n, double* a, int ia, int ja, const int desca[9], int* info);
void dlaf_pdsyevd(char uplo, int n, double* a, int ia, int ja, const int desca[9],
                  double* w, double* z, int iz, int jz, const int descz[9], int* info);
void dlaf_pzheevd(char uplo, int n, dlaf_complex_z* a, int ia, int ja, const int desca[9],
                  double* w, dlaf_complex_z* z, int iz, int jz, const int descz[9], int* info);
void dlaf_pdsygvd(char uplo, int n, double* a, int ia, int ja, const int desca[9],
                  double* b, int ib, int jb, const int descb[9],
                  double* w, double* z, int iz, int jz, const int descz[9], int* info);

#ifdef __cplusplus
}
#endif
#endif /* DLAF_C_H */
