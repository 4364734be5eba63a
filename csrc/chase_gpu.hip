// GPU bulge chase: persistent single-wave workgroups, one sweep per WG in
// round-robin, wavefront-pipelined through agent-scope flags.
//
// Mirrors csrc/band_chase.cpp's run_sweep unit-for-unit (same windows, same
// publish protocol: unit u of sweep s needs done[s-1] >= u+3), so the sweep
// windows stay disjoint by construction. Band layout identical: row-major
// [size, ld] with band[j][d] = A[j+d, j], ld >= 2b.
//
// Reference counterpart: the CPU sweep workers of
// eigensolver/band_to_tridiag/mc.h:666-693 — the reference keeps this stage
// on the host; here the band (n x 2b), the reflector store and the flags all
// stay in HBM and ~min(W, n/3b) sweeps run concurrently, one wavefront each.
//
// Cross-WG visibility uses the MI355X idiom (per-XCD L2s are not coherent;
// see MI355X_MICROARCH.md "Workgroup dispatch, XCD placement &
// inter-workgroup visibility"):
//   producer: plain stores -> __syncthreads -> lane0 fence(release, agent)
//             -> s_waitcnt vmcnt(0) -> relaxed agent flag store
//   consumer: relaxed agent poll -> fence(acquire, agent) -> __syncthreads
// Every spin is BOUNDED: on exhaustion the WG raises a global abort flag and
// all WGs drain, so a scheduling pathology surfaces as a host error, never a
// hung GPU.

#include <hip/hip_runtime.h>

#include <cstdint>

#include "cplx.h"

namespace {

constexpr int BMAX = 64;      // wave size == max band width handled per lane
constexpr int SPIN_LIMIT = 1 << 23;

typedef int v4i_ __attribute__((ext_vector_type(4)));

// Write-through (sc1) stores for the cross-WG band hand-off: the line is
// written through to memory and DROPPED from this XCD's L2, so the publish
// needs only a per-lane vmcnt drain + flag — no agent release fence writing
// back the whole dirty L2 (~80 us/hop measured in round 1, the reason the
// f64 GPU chase lost to the CPU wavefront; MI355X_MICROARCH.md
// "publish-large": 3.0 vs 8.2 us per 64 KB). §5.7: end asm with s_nop 1.
__device__ inline void store16_sc1(void* dst, v4i_ v) {
  asm volatile("global_store_dwordx4 %0, %1, off sc1\n\ts_nop 1" ::"v"(
                   (uint64_t)(uintptr_t)dst),
               "v"(v)
               : "memory");
}
__device__ inline void store8_sc1(void* dst, uint64_t v) {
  asm volatile("global_store_dwordx2 %0, %1, off sc1\n\ts_nop 1" ::"v"(
                   (uint64_t)(uintptr_t)dst),
               "v"(v)
               : "memory");
}
__device__ inline void store4_sc1(void* dst, uint32_t v) {
  asm volatile("global_store_dword %0, %1, off sc1\n\ts_nop 1" ::"v"(
                   (uint64_t)(uintptr_t)dst),
               "v"(v)
               : "memory");
}

template <typename T>
__device__ inline typename ScalarTraits<T>::real_t sabs2(T v) {
  if constexpr (sizeof(T) == sizeof(typename ScalarTraits<T>::real_t))
    return v * v;
  else
    return v.abs2();
}

template <typename R>
__device__ inline R wave_sum(R x) {
  for (int off = 32; off > 0; off >>= 1) x += __shfl_down(x, off, 64);
  return __shfl(x, 0, 64);
}

template <typename T>
__device__ inline T wave_bcast0(T x) {
  if constexpr (sizeof(T) == sizeof(typename ScalarTraits<T>::real_t))
    return __shfl(x, 0, 64);
  else
    return T{__shfl(x.re, 0, 64), __shfl(x.im, 0, 64)};
}

template <typename T>
__device__ inline T wave_sum_t(T x) {
  if constexpr (sizeof(T) == sizeof(typename ScalarTraits<T>::real_t)) {
    return wave_sum(x);
  } else {
    return T{wave_sum(x.re), wave_sum(x.im)};
  }
}

template <typename T>
__device__ inline bool is_zero(T v) {
  if constexpr (sizeof(T) == sizeof(typename ScalarTraits<T>::real_t))
    return v == T(0);
  else
    return v.re == 0 && v.im == 0;
}

// One-wave LAPACK-style larfg over lane-held x (lane p owns x_p, p < n).
// Returns tau (broadcast); x scaled in place (x_0 <- beta).
template <typename T>
__device__ T wave_reflector(int n, T& x, int lane) {
  using TR = ScalarTraits<T>;
  using R = typename TR::real_t;
  if (n <= 1) return TR::zero();
  R xn2 = wave_sum((lane >= 1 && lane < n) ? sabs2(x) : R(0));
  T alpha = wave_bcast0(x);
  R a_re = TR::real(alpha);
  R a_im;
  if constexpr (sizeof(T) == sizeof(R))
    a_im = 0;
  else
    a_im = alpha.im;
  if (xn2 == 0 && a_im == 0) return TR::zero();
  R beta = -sqrt(a_re * a_re + a_im * a_im + xn2);
  if (a_re < 0) beta = -beta;
  T tau;
  if constexpr (sizeof(T) == sizeof(R))
    tau = (beta - a_re) / beta;
  else
    tau = T((beta - a_re) / beta, -a_im / beta);
  T scale = TR::recip(alpha - TR::from_real(beta));
  if (lane >= 1 && lane < n) x = x * scale;
  if (lane == 0) x = TR::from_real(beta);
  return tau;
}

// The band window ops over the LDS-staged window W: column j+q of the band
// lives at W[q*S + d] (depth d, S = ld + pad). Serial per-lane global loads
// made the first version ~120 us/unit (latency-bound); staging the <= nn x ld
// window through LDS once per unit makes it bandwidth-bound. blk(p, q) of a
// block anchored at depth d0 is W[q*S + d0 + p - q]. Lane p owns row p
// (column q for the left apply); v/w broadcast through LDS.
// Round 2: the unit ops run on 256 THREADS (4 waves) — thread (p = tid%64,
// g = tid/64) splits the inner q/p loops 4-way with an LDS partial-sum
// reduce by wave 0. The single-wave form kept only ~one wave per CU busy
// (window LDS caps residency at 1 WG/CU), so each unit now finishes ~3x
// faster at identical numeric structure (fixed reduce order).
template <typename T>
__device__ void unit_two_sided(int nn, T tau, const T* vl, T* W, int S,
                               int tid, T* wl, T* part4) {
  using TR = ScalarTraits<T>;
  if (is_zero(tau) || nn <= 0) return;
  const int p = tid & 63;
  const int g = tid >> 6;
  T w = TR::zero();
  if (p < nn) {
    for (int q = g; q < nn; q += 4) {
      if (q < p)
        w += W[q * S + (p - q)] * vl[q];
      else if (q == p)
        w += TR::from_real(TR::real(W[p * S])) * vl[p];
      else
        w += TR::conj(W[p * S + (q - p)]) * vl[q];
    }
  }
  part4[g * BMAX + p] = w;
  __syncthreads();
  if (tid < 64) {
    const int lane = tid;
    T wp_ = part4[lane];
#pragma unroll
    for (int gg = 1; gg < 4; ++gg) wp_ += part4[gg * BMAX + lane];
    T vhu = wave_sum_t((lane < nn) ? TR::conj(vl[lane]) * wp_ : TR::zero());
    using R = typename TR::real_t;
    R t2 = sabs2(tau) / R(2);
    T half = vhu * t2;  // |tau|^2/2 * (v^H u)
    wl[lane] = (lane < nn)
                   ? tau * wp_ - half * vl[lane]
                   : TR::zero();
  }
  __syncthreads();
  if (p < nn) {
    for (int q = g; q <= p; q += 4)
      W[q * S + (p - q)] -= vl[p] * TR::conj(wl[q]) + wl[p] * TR::conj(vl[q]);
  }
  __syncthreads();
}

template <typename T>
__device__ void unit_apply_right(int m, int nn, T tau, const T* vl, T* W,
                                 int S, int d0, int tid, T* srow, T* part4) {
  using TR = ScalarTraits<T>;
  if (is_zero(tau) || m <= 0 || nn <= 0) return;
  const int p = tid & 63;
  const int g = tid >> 6;
  T sacc = TR::zero();
  if (p < m)
    for (int q = g; q < nn; q += 4) sacc += W[q * S + (d0 + p - q)] * vl[q];
  part4[g * BMAX + p] = sacc;
  __syncthreads();
  if (tid < 64) {
    T sp = part4[tid];
#pragma unroll
    for (int gg = 1; gg < 4; ++gg) sp += part4[gg * BMAX + tid];
    srow[tid] = sp * tau;
  }
  __syncthreads();
  if (p < m) {
    const T sp = srow[p];
    for (int q = g; q < nn; q += 4)
      W[q * S + (d0 + p - q)] -= sp * TR::conj(vl[q]);
  }
  __syncthreads();
}

// columns of the left apply start one band column after the window anchor:
// blk(p, q) = W[(1 + q) * S + (dL + p - q)], dL = nn - 1.
template <typename T>
__device__ void unit_apply_left(int m, int nn, T tau, const T* vl, T* W, int S,
                                int dL, int tid, T* srow, T* part4) {
  using TR = ScalarTraits<T>;
  if (is_zero(tau) || m <= 0 || nn <= 0) return;
  const int q = tid & 63;
  const int g = tid >> 6;
  T sacc = TR::zero();
  if (q < nn)
    for (int p = g; p < m; p += 4)
      sacc += TR::conj(vl[p]) * W[(1 + q) * S + (dL + p - q)];
  part4[g * BMAX + q] = sacc;
  __syncthreads();
  if (tid < 64) {
    T sq = part4[tid];
#pragma unroll
    for (int gg = 1; gg < 4; ++gg) sq += part4[gg * BMAX + tid];
    srow[tid] = sq * TR::conj(tau);
  }
  __syncthreads();
  if (q < nn) {
    const T sq = srow[q];
    for (int p = g; p < m; p += 4)
      W[(1 + q) * S + (dL + p - q)] -= sq * vl[p];
  }
  __syncthreads();
}

template <typename T>
__global__ __launch_bounds__(256) void chase_gpu_k(
    T* a, int64_t ld, int64_t size, int64_t b, T* vstore,
    const int64_t* offsets, int32_t* done, int32_t* abortf, int64_t nsweeps) {
  using TR = ScalarTraits<T>;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int64_t vstride = b + 1;
  __shared__ T vl[BMAX], wl[BMAX], part4[4 * BMAX];
  __shared__ T stau;
  __shared__ int ok_s;
  extern __shared__ char smem[];
  T* W = reinterpret_cast<T*>(smem);    // [<=b cols][S], S = ld + 2
  const int S = (int)ld + 2;

  // consumer side: bounded relaxed poll, then one agent acquire
  auto wait_flag = [&](int64_t s, int32_t need) -> bool {
    if (tid == 0) {
      int ok = 1;
      if (s > 0) {
        int spins = 0;
        while (__hip_atomic_load(&done[s - 1], __ATOMIC_RELAXED,
                                 __HIP_MEMORY_SCOPE_AGENT) < need) {
          if (++spins > SPIN_LIMIT) {
            __hip_atomic_store(abortf, 1, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT);
            ok = 0;
            break;
          }
          if ((spins & 1023) == 0 &&
              __hip_atomic_load(abortf, __ATOMIC_RELAXED,
                                __HIP_MEMORY_SCOPE_AGENT) != 0) {
            ok = 0;
            break;
          }
          // far behind -> long sleeps: idle pollers otherwise saturate the
          // fabric and starve the working wavefront
          if (spins > 64)
            __builtin_amdgcn_s_sleep(127);
          else
            __builtin_amdgcn_s_sleep(16);
        }
      }
      if (ok) __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
      ok_s = ok;
    }
    __syncthreads();
    return ok_s != 0;
  };
  // producer side: the window write-back uses write-through (sc1) stores,
  // so publishing is one per-lane vmcnt drain + barrier + relaxed flag.
  // (Valid form per MI355X_MICROARCH.md: sc1 payload -> asm vmcnt(0) ->
  // flag; the only plain cross-WG stores left are unit 0's own start
  // column, which no later sweep reads before kernel end.)
  auto publish = [&](int64_t s, int32_t val) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    if (tid == 0)
      __hip_atomic_store(&done[s], val, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
    __syncthreads();
  };

  for (int64_t s = blockIdx.x; s < nsweeps; s += gridDim.x) {
    // ---- unit 0: initial reflector of column s
    if (!wait_flag(s, 3)) return;
    const int64_t n0l = size - s - 1;
    const int n0 = (int)(n0l < b ? n0l : b);
    if (tid < 64) {
      T x = (lane < n0) ? a[1 + lane + s * ld] : TR::zero();
      T tau0 = wave_reflector(n0, x, lane);
      T* slot = vstore + offsets[s] * vstride;
      if (lane == 0) {
        slot[0] = tau0;
        stau = tau0;
      }
      T vv = (lane == 0) ? TR::from_real(1)
                         : ((lane < n0) ? x : TR::zero());
      if (lane < b) slot[1 + lane] = (lane < n0) ? vv : TR::zero();
      vl[lane] = vv;
      // band writeback: position 0 = beta, tail zeroed
      if (lane == 0 && n0 > 0) a[1 + s * ld] = x;
      if (lane >= 1 && lane < n0) a[1 + lane + s * ld] = TR::zero();
    }
    __syncthreads();
    T tau = stau;
    int32_t step = 0;
    while (true) {
      if (!wait_flag(s, (int32_t)(step + 1) + 3)) return;
      const int64_t j = 1 + s + (int64_t)step * b;
      const int64_t nnl = size - j, ml = size - b - j;
      const int nn = (int)(nnl < b ? nnl : b);
      const int m = (int)(ml < b ? ml : b);
      // stage the unit's window (cols j..j+nn-1) into LDS. Column c is only
      // touched at depths <= nn+m-c (the bulge is a lower-triangular wedge),
      // so the staircase load moves ~half of nn*ld — the cross-XCD payload
      // read is the dominant hop cost.
      {
        const T* src = a + j * ld;
        const int dmax0 = nn + (m > 0 ? m : 0) + 1;
        for (int c = 0; c < nn; ++c) {
          const int dm = min((int)ld, dmax0 - c);
          const T* sc_ = src + c * (int64_t)ld;
          T* wc = W + c * S;
          if constexpr (sizeof(T) >= 8) {
            // ld = 2b and S = ld + 2 keep both sides 16-B aligned
            constexpr int EPV = 16 / sizeof(T);
            const int nfull = dm / EPV;
            for (int t = tid; t < nfull; t += 256)
              reinterpret_cast<v4i_*>(wc)[t] =
                  reinterpret_cast<const v4i_*>(sc_)[t];
            for (int d = nfull * EPV + tid; d < dm; d += 256) wc[d] = sc_[d];
          } else {
            for (int d = tid; d < dm; d += 256) wc[d] = sc_[d];
          }
        }
      }
      __syncthreads();
      unit_two_sided(nn, tau, vl, W, S, tid, wl, part4);
      if (m > 0) unit_apply_right(m, nn, tau, vl, W, S, nn, tid, wl, part4);
      bool last = (m <= 1);
      if (!last) {
        if (tid < 64) {
          T x2 = (lane < m) ? W[nn + lane] : TR::zero();  // col j, depth nn+p
          T tau2 = wave_reflector(m, x2, lane);
          T* slot = vstore + (offsets[s] + step + 1) * vstride;
          if (lane == 0) {
            slot[0] = tau2;
            stau = tau2;
          }
          T vv = (lane == 0) ? TR::from_real(1)
                             : ((lane < m) ? x2 : TR::zero());
          if (lane < b) slot[1 + lane] = (lane < m) ? vv : TR::zero();
          vl[lane] = vv;
          if (lane == 0 && m > 0) W[nn] = x2;
          if (lane >= 1 && lane < m) W[nn + lane] = TR::zero();
        }
        ++step;
        __syncthreads();
        tau = stau;
        unit_apply_left(m, nn - 1, tau, vl, W, S, nn - 1, tid, wl, part4);
      }
      __syncthreads();
      // write the staircase back (same touched region), then publish
      {
        T* dst = a + j * ld;
        const int dmax0 = nn + (m > 0 ? m : 0) + 1;
        for (int c = 0; c < nn; ++c) {
          const int dm = min((int)ld, dmax0 - c);
          T* dc = dst + c * (int64_t)ld;
          const T* wc = W + c * S;
          if constexpr (sizeof(T) >= 8) {
            constexpr int EPV = 16 / sizeof(T);
            const int nfull = dm / EPV;
            for (int t = tid; t < nfull; t += 256)
              store16_sc1(dc + t * EPV,
                          reinterpret_cast<const v4i_*>(wc)[t]);
            for (int d = nfull * EPV + tid; d < dm; d += 256)
              store8_sc1(dc + d, *reinterpret_cast<const uint64_t*>(&wc[d]));
          } else {
            for (int d = tid; d < dm; d += 256)
              store4_sc1(dc + d, *reinterpret_cast<const uint32_t*>(&wc[d]));
          }
        }
      }
      if (last) break;
      publish(s, step);
    }
    publish(s, INT32_MAX);
  }
}

template <typename T>
void launch_chase(T* a, int64_t ld, int64_t size, int64_t b, T* vstore,
                  const int64_t* offsets, int32_t* done, int32_t* abortf,
                  hipStream_t stream) {
  int64_t nsweeps = size - 2;
  if (nsweeps <= 0) return;
  const size_t shbytes = (size_t)b * (2 * b + 2) * sizeof(T);
  if (shbytes > 65536) {
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(chase_gpu_k<T>),
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)shbytes);
  }
  int per_cu = 0;
  (void)hipOccupancyMaxActiveBlocksPerMultiprocessor(
      &per_cu, reinterpret_cast<const void*>(chase_gpu_k<T>), 256, shbytes);
  if (per_cu < 1) per_cu = 1;
  hipDeviceProp_t prop;
  int dev_ = 0;
  (void)hipGetDevice(&dev_);
  (void)hipGetDeviceProperties(&prop, dev_);
  int64_t W = (int64_t)per_cu * prop.multiProcessorCount;  // residency bound
  // concurrency bound: at most ~steps(sweep0)/3 sweeps can be in flight —
  // any extra WG is a pure poller burning fabric bandwidth
  int64_t active = (size + 3 * b - 1) / (3 * b) + 16;
  if (W > active) W = active;
  if (W > nsweeps) W = nsweeps;
  if (W < 1) W = 1;
  chase_gpu_k<T><<<dim3((uint32_t)W), dim3(256), shbytes, stream>>>(
      a, ld, size, b, vstore, offsets, done, abortf, nsweeps);
}

}  // namespace

extern "C" {

void chase_gpu_f64(double* a, int64_t ld, int64_t size, int64_t b,
                   double* vstore, const int64_t* offsets, int32_t* done,
                   int32_t* abortf, hipStream_t stream) {
  launch_chase(a, ld, size, b, vstore, offsets, done, abortf, stream);
}
void chase_gpu_f32(float* a, int64_t ld, int64_t size, int64_t b, float* vstore,
                   const int64_t* offsets, int32_t* done, int32_t* abortf,
                   hipStream_t stream) {
  launch_chase(a, ld, size, b, vstore, offsets, done, abortf, stream);
}
void chase_gpu_c128(double* a, int64_t ld, int64_t size, int64_t b,
                    double* vstore, const int64_t* offsets, int32_t* done,
                    int32_t* abortf, hipStream_t stream) {
  launch_chase(reinterpret_cast<cplx<double>*>(a), ld, size, b,
               reinterpret_cast<cplx<double>*>(vstore), offsets, done, abortf,
               stream);
}
void chase_gpu_c64(float* a, int64_t ld, int64_t size, int64_t b, float* vstore,
                   const int64_t* offsets, int32_t* done, int32_t* abortf,
                   hipStream_t stream) {
  launch_chase(reinterpret_cast<cplx<float>*>(a), ld, size, b,
               reinterpret_cast<cplx<float>*>(vstore), offsets, done, abortf,
               stream);
}

}  // extern "C"
