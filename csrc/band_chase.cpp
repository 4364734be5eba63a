// Band -> tridiagonal bulge chasing (CPU), with compact reflector recording.
//
// Counterpart of the reference's eigensolver/band_to_tridiag/mc.h
// (SweepWorker::start_sweep/do_step, mc.h:477-631): a Hermitian band matrix of
// bandwidth b (compact lower-band storage, column major, ld rows >= 2b so the
// transient bulge stays in storage) is reduced to real tridiagonal form by
// n-2 sweeps. Sweep s eliminates column s below the first subdiagonal with a
// length-<=b Householder reflector and chases the resulting bulge down with
// one reflector per b rows:
//
//   step t (j = 1 + s + t*b, n = min(b, size-j), m = min(b, size-b-j)):
//     two-sided H^H * A[j:j+n, j:j+n] * H          (Hermitian diag block)
//     right     A[j+n:j+n+m, j:j+n] * H            (off-diag block, fills it)
//     reflector from A[j+n:j+n+m, j]               (eliminates the spilled col)
//     left      H'^H * A[j+n:j+n+m, j+1:j+n]       (mix rows; bulge remains in
//                                                   cols j+1.. for LATER sweeps)
//
// Reflectors are stored compactly: slot (s, t) -> [tau, v0=1 implied, v1..],
// stride (b+1), per-sweep offsets; the back-transform applies them per sweep
// as disjoint row-block rank-1 updates (see dlaf_amd/algs/band2tridiag.py).
//
// Sequential v1 (the reference's MC backend is CPU too); the wavefront
// parallelization over sweeps (semaphore scheme, mc.h:666-693) can be added
// without changing the storage contract.

#include <torch/extension.h>

#include <atomic>
#include <cmath>
#include <complex>
#include <thread>
#include <vector>

namespace {

template <class T>
struct real_of {
  using type = T;
};
template <class R>
struct real_of<std::complex<R>> {
  using type = R;
};

template <class T>
inline typename real_of<T>::type re(const T& x) {
  if constexpr (std::is_same_v<T, std::complex<float>> ||
                std::is_same_v<T, std::complex<double>>)
    return x.real();
  else
    return x;
}

template <class T>
inline T conj_(const T& x) {
  if constexpr (std::is_same_v<T, std::complex<float>> ||
                std::is_same_v<T, std::complex<double>>)
    return std::conj(x);
  else
    return x;
}

template <class T>
inline typename real_of<T>::type abs2(const T& x) {
  if constexpr (std::is_same_v<T, std::complex<float>> ||
                std::is_same_v<T, std::complex<double>>)
    return x.real() * x.real() + x.imag() * x.imag();
  else
    return x * x;
}

// LAPACK-style larfg on x[0..n): returns tau; x[0] <- beta (real),
// x[1..] <- v tail (v0 = 1 implicit).
template <class T>
T hh_reflector(int64_t n, T* x) {
  using R = typename real_of<T>::type;
  if (n <= 1) {
    // still normalize a complex x[0]? nothing to eliminate
    return T(0);
  }
  R xnorm2 = 0;
  for (int64_t i = 1; i < n; ++i) xnorm2 += abs2(x[i]);
  T alpha = x[0];
  R a_re = re(alpha);
  R a_im = 0;
  if constexpr (std::is_same_v<T, std::complex<float>> ||
                std::is_same_v<T, std::complex<double>>)
    a_im = alpha.imag();
  if (xnorm2 == 0 && a_im == 0) return T(0);
  R beta = -std::sqrt(a_re * a_re + a_im * a_im + xnorm2);
  if (a_re < 0) beta = -beta;
  T tau;
  if constexpr (std::is_same_v<T, std::complex<float>> ||
                std::is_same_v<T, std::complex<double>>)
    tau = T((beta - a_re) / beta, -a_im / beta);
  else
    tau = (beta - a_re) / beta;
  T scale = T(1) / (alpha - T(beta));
  for (int64_t i = 1; i < n; ++i) x[i] *= scale;
  x[0] = T(beta);
  return tau;
}

// Two-sided Hermitian update of the n x n lower-band-stored block at column j:
// block(p, q) = a[(p - q) + (j + q) * ld], p >= q. w: scratch length n.
template <class T>
void apply_two_sided(int64_t nn, T tau, const T* v, T* a, int64_t ld, T* w) {
  if (tau == T(0) || nn <= 0) return;
  auto blk = [&](int64_t p, int64_t q) -> T& { return a[(p - q) + q * ld]; };
  // u = B v (Hermitian, lower stored)
  for (int64_t p = 0; p < nn; ++p) w[p] = T(0);
  for (int64_t q = 0; q < nn; ++q) {
    w[q] += T(re(blk(q, q))) * v[q];
    for (int64_t p = q + 1; p < nn; ++p) {
      w[p] += blk(p, q) * v[q];
      w[q] += conj_(blk(p, q)) * v[p];
    }
  }
  // w' = tau u - 1/2 |tau|^2 (v^H u) v
  T vhu = T(0);
  for (int64_t p = 0; p < nn; ++p) vhu += conj_(v[p]) * w[p];
  T half = T(abs2(tau) / typename real_of<T>::type(2)) * vhu;
  for (int64_t p = 0; p < nn; ++p) w[p] = tau * w[p] - half * v[p];
  // B -= v w'^H + w' v^H
  for (int64_t q = 0; q < nn; ++q)
    for (int64_t p = q; p < nn; ++p)
      blk(p, q) -= v[p] * conj_(w[q]) + w[p] * conj_(v[q]);
}

// Right-apply C <- C (I - tau v v^H) to the m x n block whose (p, q) entry is
// a[(d0 + p - q) + (j + q) * ld] (d0 = n for the off-diag block).
template <class T>
void apply_right(int64_t m, int64_t nn, T tau, const T* v, T* a, int64_t ld,
                 int64_t d0, T* w) {
  if (tau == T(0) || m <= 0 || nn <= 0) return;
  auto blk = [&](int64_t p, int64_t q) -> T& { return a[(d0 + p - q) + q * ld]; };
  for (int64_t p = 0; p < m; ++p) {
    T s = T(0);
    for (int64_t q = 0; q < nn; ++q) s += blk(p, q) * v[q];
    s *= tau;
    for (int64_t q = 0; q < nn; ++q) blk(p, q) -= s * conj_(v[q]);
  }
}

// Left-apply C <- (I - conj(tau) v v^H) C: v indexes ROWS.
template <class T>
void apply_left(int64_t m, int64_t nn, T tau, const T* v, T* a, int64_t ld,
                int64_t d0, T* w) {
  if (tau == T(0) || m <= 0 || nn <= 0) return;
  auto blk = [&](int64_t p, int64_t q) -> T& { return a[(d0 + p - q) + q * ld]; };
  T ct = conj_(tau);
  for (int64_t q = 0; q < nn; ++q) {
    T s = T(0);
    for (int64_t p = 0; p < m; ++p) s += conj_(v[p]) * blk(p, q);
    s *= ct;
    for (int64_t p = 0; p < m; ++p) blk(p, q) -= s * v[p];
  }
}

// One full sweep s (start + all chase steps). When `done` is non-null the
// sweep PIPELINES against its predecessor (wavefront parallelization, the
// reference's semaphore scheme mc.h:666-693): step t of sweep s may only run
// once sweep s-1 has completed step t+3, which keeps the two sweeps' band
// windows disjoint (gap b-1 columns).
template <class T>
void run_sweep(T* a, int64_t ld, int64_t size, int64_t b, T* vstore,
               const int64_t* offsets, int64_t s, T* v, T* w,
               std::atomic<int32_t>* done) {
  const int64_t vstride = b + 1;
  auto wait_pred = [&](int64_t t) {
    if (done && s > 0) {
      while (done[s - 1].load(std::memory_order_acquire) < t + 3)
        std::this_thread::yield();
    }
  };
  wait_pred(0);
  int64_t n0 = std::min(size - s - 1, b);
  T tau = hh_reflector(n0, a + 1 + s * ld);
  {
    T* slot = vstore + offsets[s] * vstride;
    slot[0] = tau;
    slot[1] = T(1);
    for (int64_t i = 1; i < n0; ++i) slot[1 + i] = a[1 + i + s * ld];
    for (int64_t i = n0; i < b; ++i) slot[1 + i] = T(0);
  }
  v[0] = T(1);
  for (int64_t i = 1; i < n0; ++i) v[i] = a[1 + i + s * ld];
  for (int64_t i = 1; i < n0; ++i) a[1 + i + s * ld] = T(0);

  int64_t step = 0;
  while (true) {
    wait_pred(step + 1);
    int64_t j = 1 + s + step * b;
    int64_t nn = std::min(b, size - j);
    int64_t m = std::min(b, size - b - j);
    apply_two_sided(nn, tau, v, a + j * ld, ld, w);
    if (m > 0) apply_right(m, nn, tau, v, a + j * ld, ld, nn, w);
    if (m <= 1) break;
    tau = hh_reflector(m, a + nn + j * ld);
    ++step;
    {
      T* slot = vstore + (offsets[s] + step) * vstride;
      slot[0] = tau;
      slot[1] = T(1);
      for (int64_t i = 1; i < m; ++i) slot[1 + i] = a[nn + i + j * ld];
      for (int64_t i = m; i < b; ++i) slot[1 + i] = T(0);
    }
    v[0] = T(1);
    for (int64_t i = 1; i < m; ++i) v[i] = a[nn + i + j * ld];
    for (int64_t i = 1; i < m; ++i) a[nn + i + j * ld] = T(0);
    apply_left(m, nn - 1, tau, v, a + (nn - 1) + (j + 1) * ld, ld, 0, w);
    if (done) done[s].store(step, std::memory_order_release);
  }
  if (done) done[s].store(INT32_MAX, std::memory_order_release);
}

// Sequential chase of the whole band matrix.
// band: [ld, size] column-major torch tensor (band.stride(1) == ld... we use
//   contiguous [size, ld] row-major = column-major [ld, size]).
// vstore: flat reflector storage; offsets[s] = slot index of sweep s's step 0;
//   slot k of sweep s lives at (offsets[s] + k) * (b + 1).
template <class T>
void chase_impl(T* a, int64_t ld, int64_t size, int64_t b, T* vstore,
                const int64_t* offsets, int nthreads) {
  // sweeps eliminate columns 0 .. size-3 (column size-2 is already tridiagonal)
  int64_t nsweeps = size - 2;
  if (nsweeps <= 0) return;
  if (nthreads <= 1 || nsweeps < 4) {
    std::vector<T> v(b + 1), w(b);
    for (int64_t s = 0; s < nsweeps; ++s)
      run_sweep(a, ld, size, b, vstore, offsets, s, v.data(), w.data(), nullptr);
    return;
  }
  std::vector<std::atomic<int32_t>> done(nsweeps);
  for (auto& d : done) d.store(0, std::memory_order_relaxed);
  std::vector<std::thread> pool;
  for (int tid = 0; tid < nthreads; ++tid) {
    pool.emplace_back([=, &done]() {
      std::vector<T> v(b + 1), w(b);
      for (int64_t s = tid; s < nsweeps; s += nthreads)
        run_sweep(a, ld, size, b, vstore, offsets, s, v.data(), w.data(),
                  done.data());
    });
  }
  for (auto& t : pool) t.join();
}

}  // namespace

// band: [size, ld] row-major (= column-major [ld, size]) with
//   band[j][d] = A[j+d, j], d = 0..ld-1, ld >= 2b (bulge headroom).
// vstore: [total_slots, b+1]; offsets: [size] int64 slot offsets per sweep.
void band_chase(torch::Tensor band, int64_t b, torch::Tensor vstore,
                torch::Tensor offsets, int64_t nthreads) {
  if (nthreads <= 0) {
    // measured sweet spot: 16 (64 threads ran 3x SLOWER at n=20k b=128 —
    // the shared band working set thrashes caches across too many cores)
    nthreads = std::min<int64_t>(16, std::thread::hardware_concurrency());
    if (nthreads < 1) nthreads = 1;
  }
  TORCH_CHECK(!band.is_cuda(), "band_chase is a CPU stage (as the reference)");
  TORCH_CHECK(band.is_contiguous() && vstore.is_contiguous());
  int64_t size = band.size(0);
  int64_t ld = band.size(1);
  TORCH_CHECK(ld >= 2 * b, "band storage needs >= 2b rows for the bulge");
  TORCH_CHECK(vstore.size(1) == b + 1);
  TORCH_CHECK(offsets.scalar_type() == torch::kInt64);
  auto offs = offsets.data_ptr<int64_t>();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::ComplexFloat, at::ScalarType::ComplexDouble,
      band.scalar_type(), "band_chase", [&] {
        using T = scalar_t;
        if constexpr (std::is_same_v<T, c10::complex<float>>) {
          chase_impl(reinterpret_cast<std::complex<float>*>(band.data_ptr<T>()),
                     ld, size, b,
                     reinterpret_cast<std::complex<float>*>(vstore.data_ptr<T>()),
                     offs, (int)nthreads);
        } else if constexpr (std::is_same_v<T, c10::complex<double>>) {
          chase_impl(reinterpret_cast<std::complex<double>*>(band.data_ptr<T>()),
                     ld, size, b,
                     reinterpret_cast<std::complex<double>*>(vstore.data_ptr<T>()),
                     offs, (int)nthreads);
        } else {
          chase_impl(band.data_ptr<T>(), ld, size, b, vstore.data_ptr<T>(), offs,
                     (int)nthreads);
        }
      });
}

// D&C merge deflation scan (reference dlaed2 semantics, merge.h:305-593):
// sequential pass over the SORTED (d, z); tiny rho*|z| deflates directly,
// near-equal neighbours are rotated (Givens) so one of the pair deflates.
// Returns the number of rotations written into rots (i, j, c, s quadruples).
// Runs in C++ because it is an inherently sequential O(k) scan that costed
// ~seconds as a Python loop at k ~ 16k.
int64_t dc_deflate_scan(torch::Tensor d, torch::Tensor z, double rho,
                        double tol, torch::Tensor deflated,
                        torch::Tensor rots) {
  TORCH_CHECK(!d.is_cuda() && d.scalar_type() == torch::kFloat64);
  int64_t k = d.size(0);
  double* dn = d.data_ptr<double>();
  double* zn = z.data_ptr<double>();
  bool* defl = deflated.data_ptr<bool>();
  double* rr = rots.data_ptr<double>();   // [k, 4] capacity
  int64_t nrot = 0;
  for (int64_t i = 0; i < k; ++i) defl[i] = std::abs(rho * zn[i]) <= tol;
  int64_t last = -1;
  for (int64_t i = 0; i < k; ++i) {
    if (defl[i]) continue;
    if (last >= 0 && (dn[i] - dn[last]) <= tol) {
      double zi = zn[last], zj = zn[i];
      double r = std::hypot(zi, zj);
      double c = zj / r, s = -zi / r;
      zn[i] = r;
      zn[last] = 0.0;
      double di = dn[last], dj = dn[i];
      dn[last] = di * c * c + dj * s * s;
      dn[i] = di * s * s + dj * c * c;
      rr[nrot * 4 + 0] = (double)last;
      rr[nrot * 4 + 1] = (double)i;
      rr[nrot * 4 + 2] = c;
      rr[nrot * 4 + 3] = s;
      ++nrot;
      defl[last] = true;
    }
    last = i;
  }
  return nrot;
}
