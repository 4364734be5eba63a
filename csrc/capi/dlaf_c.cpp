// libdlaf_c.so — C ABI over the dlaf_amd package via an embedded Python
// interpreter (counterpart of the reference's src/c_api/*).
//
// Caller buffers are ScaLAPACK-convention column-major with leading
// dimension ld; they are wrapped as non-owning numpy views (strides
// (itemsize, ld*itemsize)) so the Python side reads and writes the caller's
// memory directly — the flow of the reference's c_api (host matrix ->
// mirror -> algorithm -> copy back, src/c_api/eigensolver/eigensolver.h:30-73).
//
// The package is located relative to this .so (dladdr), overridable with
// DLAF_AMD_PYROOT. Multi-process grids rendezvous via the torchrun-style
// environment — see include/dlaf_c.h.

#include <pybind11/embed.h>
#include <pybind11/numpy.h>

#include <dlfcn.h>
#include <cstdlib>
#include <cstring>
#include <string>

#include "../../include/dlaf_c.h"

namespace py = pybind11;

namespace {

PyThreadState* main_tstate_ = nullptr;
py::module_* capi_ = nullptr;

std::string self_dir() {
  const char* env = std::getenv("DLAF_AMD_PYROOT");
  if (env) return env;
  Dl_info info;
  if (dladdr((void*)&dlaf_initialize, &info) && info.dli_fname) {
    std::string p(info.dli_fname);
    auto pos = p.rfind('/');
    return pos == std::string::npos ? "." : p.substr(0, pos);
  }
  return ".";
}

struct Gil {
  py::gil_scoped_acquire acq;
};

py::object desc_obj(const struct DLAF_descriptor& d);

py::object np_view(int ctx, void* ptr, const struct DLAF_descriptor& d,
                   const char* dtype) {
  Gil g;
  py::dtype dt(dtype);
  const long isz = dt.itemsize();
  // column-major RANK-LOCAL block-cyclic buffer (ScaLAPACK local panel):
  // (local_m, local_n) with stride ld; for a 1x1 grid this is the full matrix
  auto sh = capi_->attr("dlaf_local_shape")(ctx, desc_obj(d)).cast<py::tuple>();
  const long lm = sh[0].cast<long>();
  const long ln = sh[1].cast<long>();
  py::array arr(dt, {lm, ln}, {isz, (long)d.ld * isz}, ptr,
                py::str());  // base handle => non-owning view
  return arr;
}

py::object np_vec(void* ptr, long n, const char* dtype) {
  py::dtype dt(dtype);
  const long isz = dt.itemsize();
  return py::array(dt, {n}, {isz}, ptr, py::str());
}

py::object desc_obj(const struct DLAF_descriptor& d) {
  return capi_->attr("DLAF_descriptor")(d.m, d.n, d.mb, d.nb, d.isrc, d.jsrc,
                                        d.i, d.j, d.ld);
}

int run_inplace(const char* fn, int ctx, char uplo, void* a,
                const struct DLAF_descriptor& d, const char* dtype,
                const char* extra = nullptr) {
  Gil g;
  try {
    auto arr = np_view(ctx, a, d, dtype);
    if (extra)
      return capi_->attr(fn)(ctx, std::string(1, uplo), std::string(1, *extra),
                             arr, desc_obj(d)).cast<int>();
    return capi_->attr(fn)(ctx, std::string(1, uplo), arr, desc_obj(d)).cast<int>();
  } catch (const std::exception& e) {
    std::fprintf(stderr, "dlaf_c: %s failed: %s\n", fn, e.what());
    return -1;
  }
}

int run_eig(int ctx, char uplo, void* a, const struct DLAF_descriptor& da,
            void* w, void* z, const struct DLAF_descriptor& dz,
            const char* adt, const char* wdt, long il, long iu) {
  Gil g;
  try {
    auto arr = np_view(ctx, a, da, adt);
    auto zv = np_view(ctx, z, dz, adt);
    auto wv = np_vec(w, da.n, wdt);
    auto kw = py::dict();
    if (iu >= 0) {
      kw["il"] = il;
      kw["iu"] = iu;
    }
    return capi_->attr("dlaf_hermitian_eigensolver")(
        ctx, std::string(1, uplo), arr, desc_obj(da), wv, zv, desc_obj(dz),
        **kw).cast<int>();
  } catch (const std::exception& e) {
    std::fprintf(stderr, "dlaf_c: eigensolver failed: %s\n", e.what());
    return -1;
  }
}

int run_geig(int ctx, char uplo, void* a, const struct DLAF_descriptor& da,
             void* b, const struct DLAF_descriptor& db, void* w, void* z,
             const struct DLAF_descriptor& dz, const char* adt, const char* wdt,
             bool factorized) {
  Gil g;
  try {
    auto av = np_view(ctx, a, da, adt);
    auto bv = np_view(ctx, b, db, adt);
    auto zv = np_view(ctx, z, dz, adt);
    auto wv = np_vec(w, da.n, wdt);
    return capi_->attr("dlaf_hermitian_generalized_eigensolver")(
        ctx, std::string(1, uplo), av, desc_obj(da), bv, desc_obj(db), wv, zv,
        desc_obj(dz), py::arg("factorized") = factorized).cast<int>();
  } catch (const std::exception& e) {
    std::fprintf(stderr, "dlaf_c: gen eigensolver failed: %s\n", e.what());
    return -1;
  }
}

struct DLAF_descriptor from_sl(int n, const int d[9], int m = -1) {
  struct DLAF_descriptor out;
  out.m = m >= 0 ? m : d[2];
  out.n = d[3];
  out.mb = d[4];
  out.nb = d[5];
  out.isrc = d[6];
  out.jsrc = d[7];
  out.i = 1;
  out.j = 1;
  out.ld = d[8];
  (void)n;
  return out;
}

int sl_ctx_ = -1;

int sl_grid() {  // lazy local 1x1 context for the ScaLAPACK shims
  if (sl_ctx_ < 0) {
    dlaf_initialize(0, nullptr);
    sl_ctx_ = dlaf_create_grid(1, 1, 'R');
  }
  return sl_ctx_;
}

}  // namespace

extern "C" {

int dlaf_initialize(int, const char* const*) {
  if (capi_) return 0;
  if (!Py_IsInitialized()) {
    py::initialize_interpreter();
    {
      auto sys = py::module_::import("sys");
      sys.attr("path").attr("insert")(0, self_dir());
    }
    try {
      capi_ = new py::module_(py::module_::import("dlaf_amd.capi"));
    } catch (const std::exception& e) {
      std::fprintf(stderr, "dlaf_c: cannot import dlaf_amd.capi: %s\n", e.what());
      return -1;
    }
    main_tstate_ = PyEval_SaveThread();  // release GIL; calls re-acquire
  } else {
    py::gil_scoped_acquire g;
    auto sys = py::module_::import("sys");
    sys.attr("path").attr("insert")(0, self_dir());
    capi_ = new py::module_(py::module_::import("dlaf_amd.capi"));
  }
  return 0;
}

void dlaf_finalize(void) {
  // keep the interpreter alive: torch/HIP teardown from a foreign main is
  // not worth the risk, and the reference's dlaf_finalize is also a no-op
  // when pika was externally started.
}

int dlaf_create_grid(int nprow, int npcol, char order) {
  // Multi-process grids: the Python side initializes torch.distributed
  // (RCCL on GPU, gloo on CPU) from the torchrun-style launcher env
  // (RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT); nprow*npcol must equal
  // WORLD_SIZE. Reference counterpart: src/c_api/grid.cpp (MPI_Comm in,
  // CommunicatorGrid out) — the rendezvous here is the launcher env
  // instead of an MPI communicator.
  Gil g;
  try {
    return capi_->attr("dlaf_create_grid")(nprow, npcol, std::string(1, order))
        .cast<int>();
  } catch (const std::exception& e) {
    std::fprintf(stderr, "dlaf_c: dlaf_create_grid failed: %s\n", e.what());
    return -1;
  }
}

void dlaf_free_grid(int ctx) {
  Gil g;
  capi_->attr("dlaf_free_grid")(ctx);
}

#define CHOL(suf, ctype, npdt)                                                 \
  int dlaf_cholesky_factorization_##suf(int ctx, char uplo, ctype* a,          \
                                        struct DLAF_descriptor d) {            \
    return run_inplace("dlaf_cholesky_factorization", ctx, uplo, a, d, npdt);  \
  }                                                                            \
  int dlaf_inverse_from_cholesky_factor_##suf(int ctx, char uplo, ctype* a,    \
                                              struct DLAF_descriptor d) {      \
    return run_inplace("dlaf_inverse_from_cholesky_factor", ctx, uplo, a, d,   \
                       npdt);                                                  \
  }

CHOL(s, float, "float32")
CHOL(d, double, "float64")
CHOL(c, dlaf_complex_c, "complex64")
CHOL(z, dlaf_complex_z, "complex128")
#undef CHOL

int dlaf_symmetric_eigensolver_s(int ctx, char uplo, float* a,
                                 struct DLAF_descriptor da, float* w, float* z,
                                 struct DLAF_descriptor dz) {
  return run_eig(ctx, uplo, a, da, w, z, dz, "float32", "float32", 0, -1);
}
int dlaf_symmetric_eigensolver_d(int ctx, char uplo, double* a,
                                 struct DLAF_descriptor da, double* w, double* z,
                                 struct DLAF_descriptor dz) {
  return run_eig(ctx, uplo, a, da, w, z, dz, "float64", "float64", 0, -1);
}
int dlaf_hermitian_eigensolver_c(int ctx, char uplo, dlaf_complex_c* a,
                                 struct DLAF_descriptor da, float* w,
                                 dlaf_complex_c* z, struct DLAF_descriptor dz) {
  return run_eig(ctx, uplo, a, da, w, z, dz, "complex64", "float32", 0, -1);
}
int dlaf_hermitian_eigensolver_z(int ctx, char uplo, dlaf_complex_z* a,
                                 struct DLAF_descriptor da, double* w,
                                 dlaf_complex_z* z, struct DLAF_descriptor dz) {
  return run_eig(ctx, uplo, a, da, w, z, dz, "complex128", "float64", 0, -1);
}
int dlaf_symmetric_eigensolver_partial_spectrum_d(
    int ctx, char uplo, double* a, struct DLAF_descriptor da, double* w,
    double* z, struct DLAF_descriptor dz, long il, long iu) {
  return run_eig(ctx, uplo, a, da, w, z, dz, "float64", "float64", il, iu);
}
int dlaf_hermitian_eigensolver_partial_spectrum_z(
    int ctx, char uplo, dlaf_complex_z* a, struct DLAF_descriptor da, double* w,
    dlaf_complex_z* z, struct DLAF_descriptor dz, long il, long iu) {
  return run_eig(ctx, uplo, a, da, w, z, dz, "complex128", "float64", il, iu);
}

int dlaf_symmetric_generalized_eigensolver_d(
    int ctx, char uplo, double* a, struct DLAF_descriptor da, double* b,
    struct DLAF_descriptor db, double* w, double* z, struct DLAF_descriptor dz) {
  return run_geig(ctx, uplo, a, da, b, db, w, z, dz, "float64", "float64", false);
}
int dlaf_symmetric_generalized_eigensolver_factorized_d(
    int ctx, char uplo, double* a, struct DLAF_descriptor da, double* b,
    struct DLAF_descriptor db, double* w, double* z, struct DLAF_descriptor dz) {
  return run_geig(ctx, uplo, a, da, b, db, w, z, dz, "float64", "float64", true);
}
int dlaf_hermitian_generalized_eigensolver_z(
    int ctx, char uplo, dlaf_complex_z* a, struct DLAF_descriptor da,
    dlaf_complex_z* b, struct DLAF_descriptor db, double* w, dlaf_complex_z* z,
    struct DLAF_descriptor dz) {
  return run_geig(ctx, uplo, a, da, b, db, w, z, dz, "complex128", "float64",
                  false);
}
int dlaf_hermitian_generalized_eigensolver_factorized_z(
    int ctx, char uplo, dlaf_complex_z* a, struct DLAF_descriptor da,
    dlaf_complex_z* b, struct DLAF_descriptor db, double* w, dlaf_complex_z* z,
    struct DLAF_descriptor dz) {
  return run_geig(ctx, uplo, a, da, b, db, w, z, dz, "complex128", "float64",
                  true);
}

#define SL_POTRF(name, suf, ctype)                                             \
  void name(char uplo, int n, ctype* a, int ia, int ja, const int desca[9],    \
            int* info) {                                                       \
    (void)ia;                                                                  \
    (void)ja;                                                                  \
    *info = dlaf_cholesky_factorization_##suf(sl_grid(), uplo, a,              \
                                              from_sl(n, desca));              \
  }

SL_POTRF(dlaf_pdpotrf, d, double)
SL_POTRF(dlaf_pspotrf, s, float)
SL_POTRF(dlaf_pzpotrf, z, dlaf_complex_z)
SL_POTRF(dlaf_pcpotrf, c, dlaf_complex_c)
#undef SL_POTRF

void dlaf_pdpotri(char uplo, int n, double* a, int ia, int ja,
                  const int desca[9], int* info) {
  (void)ia;
  (void)ja;
  *info = dlaf_inverse_from_cholesky_factor_d(sl_grid(), uplo, a,
                                              from_sl(n, desca));
}

void dlaf_pdsyevd(char uplo, int n, double* a, int ia, int ja,
                  const int desca[9], double* w, double* z, int iz, int jz,
                  const int descz[9], int* info) {
  (void)ia; (void)ja; (void)iz; (void)jz;
  *info = dlaf_symmetric_eigensolver_d(sl_grid(), uplo, a, from_sl(n, desca), w,
                                       z, from_sl(n, descz));
}

void dlaf_pzheevd(char uplo, int n, dlaf_complex_z* a, int ia, int ja,
                  const int desca[9], double* w, dlaf_complex_z* z, int iz,
                  int jz, const int descz[9], int* info) {
  (void)ia; (void)ja; (void)iz; (void)jz;
  *info = dlaf_hermitian_eigensolver_z(sl_grid(), uplo, a, from_sl(n, desca), w,
                                       z, from_sl(n, descz));
}

void dlaf_pdsygvd(char uplo, int n, double* a, int ia, int ja,
                  const int desca[9], double* b, int ib, int jb,
                  const int descb[9], double* w, double* z, int iz, int jz,
                  const int descz[9], int* info) {
  (void)ia; (void)ja; (void)ib; (void)jb; (void)iz; (void)jz;
  *info = dlaf_symmetric_generalized_eigensolver_d(
      sl_grid(), uplo, a, from_sl(n, desca), b, from_sl(n, descb), w, z,
      from_sl(n, descz));
}

}  // extern "C"
