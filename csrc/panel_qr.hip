// Cooperative panel QR for the reduction-to-band panel factorization.
//
// One kernel launch factors an entire m x nb panel (LAPACK geqrf convention:
// R in the upper triangle, reflector tails below the diagonal, taus out).
// The host-driven equivalent costs ~20 kernel launches per column and
// dominates reduction_to_band wall time (measured 13.3 s of a 27 s SYEV
// n=16384); here the column loop runs inside the kernel with cooperative
// grid syncs (3 per column).
//
// Per column j:
//   A: partial sums |x_tail|^2 and w_raw[q] = x_tail^H P[tail,q] (atomics into
//      a per-column workspace slice); stage row j (wrow[q] = P[j,q]) and the
//      diagonal alpha so later phases never race on live P entries;
//   B: every thread re-derives beta/tau/scale from the reduced sums and
//      applies  P[i,q] -= tau * v_i * (wrow[q] + conj(scale) w_raw[q]),
//      with v_i read from the UNSCALED stored tail (v_i = x_i * scale);
//   C: scale the stored tail, write beta and tau.
//
// Reference counterpart: the CPU thread-team panel factorization of
// eigensolver/reduction_to_band/impl.h:252-350 (+ per-column larfg).

#include <hip/hip_runtime.h>
#include <hip/hip_cooperative_groups.h>

#include "cplx.h"

namespace cg = cooperative_groups;

namespace {

// Grid barrier: sense-reversing device-scope counter with agent-scope
// release/acquire fences (MI355X_MICROARCH.md "barrier-counter": 7.4 us
// host-paired at 1 WG/CU vs 26.3 us for cooperative_groups::sync at 256
// WGs — the cg sync is software on ROCm 7.2 and was ~60% of panel_qr's
// per-panel time). Requires co-resident blocks (the cooperative launch
// below still performs the residency check).
__device__ inline void grid_barrier(int32_t* cnt, int32_t* gen, int nblocks) {
  __syncthreads();
  if (threadIdx.x == 0) {
    const int32_t g = __hip_atomic_load(gen, __ATOMIC_RELAXED,
                                        __HIP_MEMORY_SCOPE_AGENT);
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    const int32_t old = __hip_atomic_fetch_add(cnt, 1, __ATOMIC_RELAXED,
                                               __HIP_MEMORY_SCOPE_AGENT);
    if (old == nblocks - 1) {
      __hip_atomic_store(cnt, 0, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      __hip_atomic_store(gen, g + 1, __ATOMIC_RELEASE,
                         __HIP_MEMORY_SCOPE_AGENT);
    } else {
      while (__hip_atomic_load(gen, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT) == g)
        __builtin_amdgcn_s_sleep(8);
    }
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  }
  __syncthreads();
}

template <class T>
struct RealOf {
  using type = T;
};
template <>
struct RealOf<cplx<double>> {
  using type = double;
};
template <>
struct RealOf<cplx<float>> {
  using type = float;
};

template <class T>
__device__ inline T conjv(T x) {
  return x;
}
__device__ inline cplx<double> conjv(cplx<double> x) { return {x.re, -x.im}; }
__device__ inline cplx<float> conjv(cplx<float> x) { return {x.re, -x.im}; }

template <class T>
__device__ inline typename RealOf<T>::type abs2v(T x) {
  return x * x;
}
__device__ inline double abs2v(cplx<double> x) { return x.re * x.re + x.im * x.im; }
__device__ inline float abs2v(cplx<float> x) { return x.re * x.re + x.im * x.im; }

__device__ inline void atomic_addT(double* p, double v) { atomicAdd(p, v); }
__device__ inline void atomic_addT(float* p, float v) { atomicAdd(p, v); }
__device__ inline void atomic_addT(cplx<double>* p, cplx<double> v) {
  atomicAdd(&reinterpret_cast<double*>(p)[0], v.re);
  atomicAdd(&reinterpret_cast<double*>(p)[1], v.im);
}
__device__ inline void atomic_addT(cplx<float>* p, cplx<float> v) {
  atomicAdd(&reinterpret_cast<float*>(p)[0], v.re);
  atomicAdd(&reinterpret_cast<float*>(p)[1], v.im);
}

template <class T, class R>
__device__ inline void derive(R xn2, T alpha, T& tau, T& scale, R& beta,
                              bool& degen) {
  R a_re, a_im;
  if constexpr (std::is_same_v<T, cplx<double>> || std::is_same_v<T, cplx<float>>) {
    a_re = alpha.re;
    a_im = alpha.im;
  } else {
    a_re = alpha;
    a_im = R(0);
  }
  degen = (xn2 == R(0)) && (a_im == R(0));
  if (degen) {
    tau = ScalarTraits<T>::zero();
    scale = ScalarTraits<T>::from_real(R(1));
    beta = a_re;
    return;
  }
  beta = -sqrt(a_re * a_re + a_im * a_im + xn2);
  if (a_re < 0) beta = -beta;
  if constexpr (std::is_same_v<T, cplx<double>> || std::is_same_v<T, cplx<float>>) {
    tau = T{(beta - a_re) / beta, -a_im / beta};
    T denom = T{a_re - beta, a_im};
    R d2 = abs2v(denom);
    scale = T{denom.re / d2, -denom.im / d2};
  } else {
    tau = (beta - a_re) / beta;
    scale = R(1) / (a_re - beta);
  }
}

// Workspace: norms[nb] (real, zeroed); wraw[3 * nb * nb] (T, zeroed):
// row j holds the w_raw sums (parity-half 0), row nb + j stages {alpha at
// [j], P[j,q] at [q]}, row 2nb + j the parity-half-1 sums (phase A runs TWO
// reducer blocks per column — one per row parity — so the 128-block grid is
// not half idle on a <=64-column panel).
template <class T>
__global__ void panel_qr_kernel(T* __restrict__ P, long m, int nb, long ldp,
                                T* __restrict__ taus,
                                typename RealOf<T>::type* __restrict__ norms,
                                T* __restrict__ wraw, int32_t* __restrict__ sync_ws) {
  using R = typename RealOf<T>::type;
  const int nblocks = (int)gridDim.x;
  const int tid = threadIdx.x;
  const int nthreads = blockDim.x;
  const long gstride = (long)gridDim.x * nthreads;
  const long gid0 = (long)blockIdx.x * nthreads + tid;

  const int ncols = nb < m ? nb : (int)m;
  for (int j = 0; j < ncols; ++j) {
    T* wj = wraw + (long)j * nb;
    T* wrow = wraw + (long)(nb + j) * nb;
    T* whalf = wraw + (long)(2 * nb + j) * nb;
    // ---- phase A: each block OWNS whole output values (no cross-block
    // atomics — an atomics formulation serializes on <= nb addresses and
    // measured 48 ms/panel). Block task bq covers column q = j + bq/2,
    // row-parity half h = bq & 1; h == 0 writes wj/norms, h == 1 whalf
    // (summed in phase B). q == j computes the column norm instead. ----
    {
      __shared__ R sredr[256];
      __shared__ T sredt[256];
      const int ntasks = 2 * (nb - j);
      for (int bq = (int)blockIdx.x; bq < ntasks; bq += gridDim.x) {
        const int q = j + (bq >> 1);
        const int h = bq & 1;
        if (q == j) {
          R part = R(0);
          for (long i = j + 1 + h + 2 * (long)tid; i < m; i += 2 * nthreads)
            part += abs2v(P[i * ldp + j]);
          sredr[tid] = part;
          __syncthreads();
          for (int s = nthreads / 2; s > 0; s >>= 1) {
            if (tid < s) sredr[tid] += sredr[tid + s];
            __syncthreads();
          }
          if (tid == 0) {
            if (h == 0)
              norms[j] = sredr[0];
            else
              whalf[j] = ScalarTraits<T>::from_real(sredr[0]);
          }
        } else {
          T part = ScalarTraits<T>::zero();
          for (long i = j + 1 + h + 2 * (long)tid; i < m; i += 2 * nthreads)
            part += conjv(P[i * ldp + j]) * P[i * ldp + q];
          sredt[tid] = part;
          __syncthreads();
          for (int s = nthreads / 2; s > 0; s >>= 1) {
            if (tid < s) sredt[tid] += sredt[tid + s];
            __syncthreads();
          }
          if (tid == 0) {
            if (h == 0)
              wj[q] = sredt[0];
            else
              whalf[q] = sredt[0];
          }
        }
        __syncthreads();
      }
      // stage row j and the diagonal
      for (long q = j + gid0; q < nb; q += gstride) wrow[q] = P[(long)j * ldp + q];
    }
    grid_barrier(sync_ws, sync_ws + 1, nblocks);
    // ---- phase B: trailing update (reads only staged row/diag + unscaled x).
    // Tail scaling, beta and tau writes are DEFERRED to the epilogue: column
    // j is never read again by later columns, so leaving its tail unscaled
    // halves the grid syncs (2 per column).
    {
      T tau, scale;
      R beta;
      bool degen;
      derive<T, R>(norms[j] + ScalarTraits<T>::real(whalf[j]), wrow[j], tau,
                   scale, beta, degen);
      if (!degen) {
        T ctau = conjv(tau);
        T cscale = conjv(scale);
        for (long idx = gid0; idx < (m - j) * (long)(nb - j - 1); idx += gstride) {
          long i = j + idx / (nb - j - 1);
          int q = j + 1 + (int)(idx % (nb - j - 1));
          T wq = wrow[q] + cscale * (wj[q] + whalf[q]);
          T vi = (i == j) ? ScalarTraits<T>::from_real(R(1))
                          : P[i * ldp + j] * scale;
          P[i * ldp + q] = P[i * ldp + q] - ctau * vi * wq;
        }
      }
    }
    grid_barrier(sync_ws, sync_ws + 1, nblocks);
  }
  // ---- epilogue: per-column tail scaling + beta/tau writes ----
  for (int j = 0; j < ncols; ++j) {
    T* wrow = wraw + (long)(nb + j) * nb;
    T* whalf = wraw + (long)(2 * nb + j) * nb;
    T tau, scale;
    R beta;
    bool degen;
    derive<T, R>(norms[j] + ScalarTraits<T>::real(whalf[j]), wrow[j], tau,
                 scale, beta, degen);
    if (!degen) {
      for (long i = j + 1 + gid0; i < m; i += gstride)
        P[i * ldp + j] = P[i * ldp + j] * scale;
    }
    if (blockIdx.x == 0 && tid == 0) {
      taus[j] = tau;
      if (!degen) P[(long)j * ldp + j] = ScalarTraits<T>::from_real(beta);
    }
  }
}

}  // namespace

namespace {
// {cnt, gen} pair for the grid barrier; gen persists across launches (the
// sense-reversing protocol only needs cnt == 0 at entry, which every
// completed barrier leaves behind)
int32_t* sync_workspace() {
  static int32_t* ws = [] {
    int32_t* p = nullptr;
    if (hipMalloc(&p, 2 * sizeof(int32_t)) != hipSuccess) return (int32_t*)nullptr;
    (void)hipMemset(p, 0, 2 * sizeof(int32_t));
    return p;
  }();
  return ws;
}
}  // namespace

extern "C" {

// Launch helper; returns 0 on success. norms/wraw must be zeroed by caller.
#define DEF_PANEL_QR(SUF, T, R)                                               \
  int panel_qr_##SUF(T* P, long m, int nb, long ldp, T* taus, R* norms,       \
                     T* wraw, hipStream_t stream) {                           \
    int threads = 256;                                                        \
    hipDeviceProp_t prop;                                                     \
    int dev_ = 0;                                                             \
    (void)hipGetDevice(&dev_);                                                \
    (void)hipGetDeviceProperties(&prop, dev_);                                \
    int per_cu = 0;                                                           \
    (void)hipOccupancyMaxActiveBlocksPerMultiprocessor(                       \
        &per_cu, reinterpret_cast<const void*>(&panel_qr_kernel<T>), threads, \
        0);                                                                   \
    int blocks = prop.multiProcessorCount * (per_cu > 0 ? per_cu : 1);        \
    /* cap chosen when the cg grid.sync cost grew with WG count; the agent  \
       counter barrier is cheap at 256 WGs so a higher cap may now win —    \
       tunable via DLAF_PANEL_QR_BLOCKS (256 untested at scale)             */ \
    static const int cap = [] {                                               \
      const char* v = getenv("DLAF_PANEL_QR_BLOCKS");                         \
      return v ? atoi(v) : 128;                                               \
    }();                                                                      \
    if (blocks > cap) blocks = cap;                                           \
    int32_t* sync_ws = sync_workspace();                                      \
    if (!sync_ws) return (int)hipErrorOutOfMemory;                            \
    void* args[] = {(void*)&P,    (void*)&m,     (void*)&nb,  (void*)&ldp,    \
                    (void*)&taus, (void*)&norms, (void*)&wraw,                \
                    (void*)&sync_ws};                                         \
    hipError_t err = hipLaunchCooperativeKernel(                              \
        reinterpret_cast<void*>(&panel_qr_kernel<T>), dim3(blocks),           \
        dim3(threads), args, 0, stream);                                      \
    return err == hipSuccess ? 0 : (int)err;                                  \
  }

DEF_PANEL_QR(f64, double, double)
DEF_PANEL_QR(f32, float, float)
DEF_PANEL_QR(c128, cplx<double>, double)
DEF_PANEL_QR(c64, cplx<float>, float)

}  // extern "C"
