// Secular-equation root solver for the D&C merge, one kernel launch.
//
// Roots of f(lam) = 1 + rho * sum_i z_i^2 / (d_i - lam), rho > 0, d ascending.
// One THREAD per root (the reference runs per-root laed4 in host thread
// teams, merge.h:813-900; the torch formulation costs ~25 elementwise
// launches per iteration — ~300k launches per full solve). Each thread:
//   1. picks the shift pole (left/right by the sign of f at the interval
//      midpoint) — the (pole, offset) representation keeps d_i - lam exact;
//   2. runs the laed4-style two-pole rational iteration with bracketing;
//   3. polishes with Illinois regula falsi (guaranteed bracket shrinkage).
// Per-thread convergence exit; poles/weights are read sequentially per
// f-evaluation (L2-broadcast friendly).

#include <hip/hip_runtime.h>

#include <cmath>

namespace {

template <class T>
__device__ inline T eval_f(const T* __restrict__ dd, const T* __restrict__ z2,
                           int k, T rho, const T* __restrict__ delta0_base,
                           T shift, T mu) {
  // f = 1 + rho * sum z2_i / ((d_i - shift) - mu)
  T s = T(1);
  for (int i = 0; i < k; ++i) s += rho * z2[i] / ((dd[i] - shift) - mu);
  return s;
}

template <class T>
__global__ void secular_kernel(const T* __restrict__ d,
                               const T* __restrict__ z2, int k, T rho,
                               long long* __restrict__ sidx,
                               T* __restrict__ mu_out) {
  const int j = blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= k) return;
  const T dj = d[j];
  const T dj1 = (j == k - 1) ? dj + rho : d[j + 1];
  const T mid = T(0.5) * (dj + dj1);
  // shift decision
  T fm = T(1);
  for (int i = 0; i < k; ++i) fm += rho * z2[i] / (d[i] - mid);
  const bool leftp = (fm >= T(0)) || (j == k - 1);
  const int p = leftp ? j : j + 1;
  const T shift = d[p];
  T lo = leftp ? T(0) : mid - shift;
  T hi = leftp ? mid - shift : T(0);
  if (j == k - 1) {
    lo = T(0);
    hi = rho;
  }
  const T d1_0 = dj - shift;    // left pole in mu coords (before -mu)
  const T d2_0 = dj1 - shift;

  T mu = T(0.5) * (lo + hi);
  // ---- rational iteration ----
  for (int it = 0; it < 40; ++it) {
    T f = T(1), psi_p = T(0), phi_p = T(0);
    for (int i = 0; i < k; ++i) {
      const T diff = (d[i] - shift) - mu;
      const T t = z2[i] / diff;
      f += rho * t;
      const T t2 = t / diff;
      if (i <= j)
        psi_p += t2;
      else
        phi_p += t2;
    }
    const T d1 = d1_0 - mu;
    const T d2 = d2_0 - mu;
    const T P = rho * psi_p * d1 * d1;
    const T Q = rho * phi_p * d2 * d2;
    const T c = f - rho * psi_p * d1 - rho * phi_p * d2;
    T s;
    if (j == k - 1) {
      s = d1 + P / (fabs(c) < T(1e-300) ? T(1e-300) : c);
    } else {
      const T a = c;
      const T b = -(c * (d1 + d2) + P + Q);
      const T c2 = c * d1 * d2 + P * d2 + Q * d1;
      T disc = b * b - T(4) * a * c2;
      disc = disc > T(0) ? sqrt(disc) : T(0);
      const T qq = T(-0.5) * (b + (b >= T(0) ? disc : -disc));
      const T r1 = qq / (fabs(a) < T(1e-300) ? T(1e-300) : a);
      const T r2 = c2 / (fabs(qq) < T(1e-300) ? T(1e-300) : qq);
      s = (r1 > d1 && r1 < d2) ? r1 : r2;
    }
    if (f < T(0))
      lo = mu;
    else
      hi = mu;
    T mu_n = mu + s;
    if (!isfinite(mu_n) || mu_n <= lo || mu_n >= hi) mu_n = T(0.5) * (lo + hi);
    // convergence: step negligible relative to the offset
    if (fabs(mu_n - mu) <= T(1e-16) * (fabs(mu_n) + T(1e-300))) {
      mu = mu_n;
      break;
    }
    mu = mu_n;
  }
  // ---- Illinois polish ----
  T flo = -T(1e300), fhi = T(1e300);
  {
    T f = T(1);
    for (int i = 0; i < k; ++i) f += rho * z2[i] / ((d[i] - shift) - mu);
    if (f < T(0)) {
      lo = mu;
      flo = f;
    } else {
      hi = mu;
      fhi = f;
    }
  }
  int side = 0;
  for (int it = 0; it < 24; ++it) {
    const T den = fhi - flo;
    T x = (fabs(den) > T(0)) ? (lo * fhi - hi * flo) / den : T(0.5) * (lo + hi);
    if (!(x > lo && x < hi) || !isfinite(x)) x = T(0.5) * (lo + hi);
    T fx = T(1);
    for (int i = 0; i < k; ++i) fx += rho * z2[i] / ((d[i] - shift) - x);
    if (fx < T(0)) {
      if (side < 0) fhi *= T(0.5);
      lo = x;
      flo = fx;
      side = -1;
    } else {
      if (side > 0) flo *= T(0.5);
      hi = x;
      fhi = fx;
      side = 1;
    }
    if (fx == T(0)) break;
    // converged: bracket width at relative machine precision
    if (hi - lo <= T(1e-16) * (fabs(lo) + fabs(hi) + T(1e-300))) break;
  }
  mu = (fabs(flo) < fabs(fhi)) ? lo : hi;
  sidx[j] = p;
  mu_out[j] = mu;
}

// ---- wave-per-root variant (round 2) ----
// One 64-lane WAVE per root: the pole sums go lane-strided (coalesced d/z2
// reads) with wave reductions; the scalar iteration state is computed
// redundantly on every lane from the broadcast sums, so all branches stay
// wave-uniform. The thread-per-root kernel above keeps small merges (its
// grid is k/128 workgroups — at k = 8192 that busied 32 of 256 CUs and
// made secular the largest kernel of the whole SYEV profile).

__device__ inline double wsum64(double x) {
  for (int off = 32; off > 0; off >>= 1) x += __shfl_down(x, off, 64);
  return __shfl(x, 0, 64);
}

template <class T>
__global__ __launch_bounds__(256) void secular_wave_kernel(
    const T* __restrict__ d, const T* __restrict__ z2, int k, T rho,
    long long* __restrict__ sidx, T* __restrict__ mu_out) {
  const int lane = threadIdx.x & 63;
  const int j = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (j >= k) return;
  const T dj = d[j];
  const T dj1 = (j == k - 1) ? dj + rho : d[j + 1];
  const T mid = T(0.5) * (dj + dj1);
  T part = T(0);
  for (int i = lane; i < k; i += 64) part += rho * z2[i] / (d[i] - mid);
  const T fm = T(1) + wsum64(part);
  const bool leftp = (fm >= T(0)) || (j == k - 1);
  const int p = leftp ? j : j + 1;
  const T shift = d[p];
  T lo = leftp ? T(0) : mid - shift;
  T hi = leftp ? mid - shift : T(0);
  if (j == k - 1) {
    lo = T(0);
    hi = rho;
  }
  const T d1_0 = dj - shift;
  const T d2_0 = dj1 - shift;

  T mu = T(0.5) * (lo + hi);
  for (int it = 0; it < 40; ++it) {
    T pf = T(0), ppsi = T(0), pphi = T(0);
    for (int i = lane; i < k; i += 64) {
      const T diff = (d[i] - shift) - mu;
      const T t = z2[i] / diff;
      pf += t;
      const T t2 = t / diff;
      if (i <= j)
        ppsi += t2;
      else
        pphi += t2;
    }
    const T f = T(1) + rho * wsum64(pf);
    const T psi_p = wsum64(ppsi);
    const T phi_p = wsum64(pphi);
    const T d1 = d1_0 - mu;
    const T d2 = d2_0 - mu;
    const T P = rho * psi_p * d1 * d1;
    const T Q = rho * phi_p * d2 * d2;
    const T c = f - rho * psi_p * d1 - rho * phi_p * d2;
    T s;
    if (j == k - 1) {
      s = d1 + P / (fabs(c) < T(1e-300) ? T(1e-300) : c);
    } else {
      const T a = c;
      const T b = -(c * (d1 + d2) + P + Q);
      const T c2 = c * d1 * d2 + P * d2 + Q * d1;
      T disc = b * b - T(4) * a * c2;
      disc = disc > T(0) ? sqrt(disc) : T(0);
      const T qq = T(-0.5) * (b + (b >= T(0) ? disc : -disc));
      const T r1 = qq / (fabs(a) < T(1e-300) ? T(1e-300) : a);
      const T r2 = c2 / (fabs(qq) < T(1e-300) ? T(1e-300) : qq);
      s = (r1 > d1 && r1 < d2) ? r1 : r2;
    }
    if (f < T(0))
      lo = mu;
    else
      hi = mu;
    T mu_n = mu + s;
    if (!isfinite(mu_n) || mu_n <= lo || mu_n >= hi) mu_n = T(0.5) * (lo + hi);
    if (fabs(mu_n - mu) <= T(1e-16) * (fabs(mu_n) + T(1e-300))) {
      mu = mu_n;
      break;
    }
    mu = mu_n;
  }
  T flo = -T(1e300), fhi = T(1e300);
  {
    T pf = T(0);
    for (int i = lane; i < k; i += 64)
      pf += z2[i] / ((d[i] - shift) - mu);
    const T f = T(1) + rho * wsum64(pf);
    if (f < T(0)) {
      lo = mu;
      flo = f;
    } else {
      hi = mu;
      fhi = f;
    }
  }
  int side = 0;
  for (int it = 0; it < 24; ++it) {
    const T den = fhi - flo;
    T x = (fabs(den) > T(0)) ? (lo * fhi - hi * flo) / den : T(0.5) * (lo + hi);
    if (!(x > lo && x < hi) || !isfinite(x)) x = T(0.5) * (lo + hi);
    T pf = T(0);
    for (int i = lane; i < k; i += 64)
      pf += z2[i] / ((d[i] - shift) - x);
    const T fx = T(1) + rho * wsum64(pf);
    if (fx < T(0)) {
      if (side < 0) fhi *= T(0.5);
      lo = x;
      flo = fx;
      side = -1;
    } else {
      if (side > 0) flo *= T(0.5);
      hi = x;
      fhi = fx;
      side = 1;
    }
    if (fx == T(0)) break;
    if (hi - lo <= T(1e-16) * (fabs(lo) + fabs(hi) + T(1e-300))) break;
  }
  mu = (fabs(flo) < fabs(fhi)) ? lo : hi;
  if (lane == 0) {
    sidx[j] = p;
    mu_out[j] = mu;
  }
}

}  // namespace

extern "C" {

void secular_roots_f64(const double* d, const double* z2, int k, double rho,
                       long long* sidx, double* mu, hipStream_t stream) {
  if (k <= 0) return;
  if (k >= 128) {
    // wave per root: 4 roots per 256-thread workgroup (a thread-per-root
    // grid at k = 512 is 4 workgroups on a 256-CU chip)
    const int blocks = (k + 3) / 4;
    secular_wave_kernel<double><<<blocks, 256, 0, stream>>>(d, z2, k, rho,
                                                            sidx, mu);
    return;
  }
  const int threads = 128;
  const int blocks = (k + threads - 1) / threads;
  secular_kernel<double><<<blocks, threads, 0, stream>>>(d, z2, k, rho, sidx, mu);
}

}  // extern "C"
