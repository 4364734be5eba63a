// Band-to-tridiag back-transform: one kernel launch per sweep GROUP.
//
// The compact-WY window chain of bt_band_to_tridiag (reference
// eigensolver/bt_band_to_tridiag/impl.h:59-1031) is sequential in k (windows
// overlap by G-1 rows) but INDEPENDENT across eigenvector columns. The torch
// formulation paid ~3 GEMM launches per window (~70k launches per solve,
// host-launch bound). Here each 256-thread workgroup owns a CW=16-column
// slice of E and marches the ENTIRE chain of one group in LDS:
//
//   per window k:  W1 = V_k^H seg   (MFMA, K = R ring rows)
//                  seg -= (V_k T_k) W1        (MFMA, K = G)
//                  store the b finalized rows; load the next b rows
//
// The E slice lives in an LDS ring of R = ceil((G+b-1)/b)*b rows (the b-row
// shift between windows makes slot reuse exact: the rows finalized by
// window k free the slots window k+1's fresh rows need). V (padded to R
// rows) and VT^T = (V T)^T (padded to R cols) stream from L2 — every
// workgroup reads the same window panels. Requirements (else the torch path
// runs): G % 32 == 0, b % 16 == 0, f64/c128.
#include <hip/hip_runtime.h>

#include <cstdint>

#include "cplx.h"

namespace {

typedef double v4d __attribute__((ext_vector_type(4)));

constexpr int CW = 16;   // eigenvector columns per workgroup
constexpr int CWP = CW + 2;

template <typename T>
struct BtMfma;
template <>
struct BtMfma<double> {
  static __device__ inline v4d mma(double a, double b, v4d c) {
    return __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, c, 0, 0, 0);
  }
};

// real f64 ---------------------------------------------------------------

__global__ __launch_bounds__(256) void bt_group_f64(
    double* __restrict__ E, int64_t nE, int64_t npad, const double* __restrict__ V,
    const double* __restrict__ VTt, int64_t base0, int b, int G, int R,
    int nwin) {
  extern __shared__ double S[];  // ring[R][CWP] then W1[G][CWP]
  double* ring = S;
  double* W1 = S + (int64_t)R * CWP;

  const int tid = threadIdx.x;
  const int lane = tid & 63, w = tid >> 6;
  const int li = lane & 15, lk = lane >> 4;
  const int64_t col0 = (int64_t)blockIdx.x * CW;

  auto ld_row = [&](int64_t grow, int slot) {
    // one row (CW cols) per 16 threads; caller loops rows
    const int c = tid % CW;
    const int r = tid / CW;  // 0..15 rows per pass
    (void)r;
    double v = 0.0;
    if (grow < npad && col0 + c < nE) v = E[grow * nE + col0 + c];
    ring[slot * CWP + c] = v;
  };
  (void)ld_row;

  // bulk row mover: rows [r0, r0+cnt) of the window frame (frame row f ->
  // global row base + f, ring slot (sbase + f) % R)
  auto move_rows = [&](int64_t gbase, int sbase, int f0, int cnt, bool store) {
    for (int f = f0 + tid / CW; f < f0 + cnt; f += 256 / CW) {
      const int c = tid % CW;
      int slot = sbase + f;
      if (slot >= R) slot -= R;
      if (slot >= R) slot -= R;
      const int64_t grow = gbase + f;
      if (store) {
        if (grow < npad && col0 + c < nE)
          E[grow * nE + col0 + c] = ring[slot * CWP + c];
      } else {
        double v = 0.0;
        if (grow < npad && col0 + c < nE) v = E[grow * nE + col0 + c];
        ring[slot * CWP + c] = v;
      }
    }
  };

  // prologue: frame of window 0 is rows [base0, base0 + R); load the first
  // R - b rows (each window loads its last b rows at the end of step k-1)
  move_rows(base0, 0, 0, R - b, false);
  __syncthreads();

  for (int k = 0; k < nwin; ++k) {
    const int64_t gbase = base0 + (int64_t)k * b;
    const int sbase = (int)(((int64_t)k * b) % R);
    // load the last b rows of this window's frame
    move_rows(gbase, sbase, R - b, b, false);
    __syncthreads();

    const double* Vk = V + (int64_t)k * R * G;
    const double* VTk = VTt + (int64_t)k * G * R;

    // W1[g][c] = sum_h conj(V[h][g]) * ring[h][c]. All of this wave's
    // fragments run in ONE k-loop (independent accumulators interleave so
    // the MFMA dependent-accumulator latency never serializes), and the
    // ring read is shared across fragments.
    {
      constexpr int NF = 2;  // G==128: (G/16)/4 frags per wave
      v4d acc[NF] = {{0, 0, 0, 0}, {0, 0, 0, 0}};
#pragma unroll 4
      for (int h0 = 0; h0 < R; h0 += 4) {
        const int h = h0 + lk;
        int slot = sbase + h;
        if (slot >= R) slot -= R;
        const double bb = ring[slot * CWP + li];
#pragma unroll
        for (int j = 0; j < NF; ++j) {
          const int f = w + 4 * j;
          if (f < G / 16) {
            const double a = Vk[(int64_t)h * G + f * 16 + li];
            acc[j] = BtMfma<double>::mma(a, bb, acc[j]);
          }
        }
      }
#pragma unroll
      for (int j = 0; j < NF; ++j)
        if (w + 4 * j < G / 16)
#pragma unroll
          for (int r = 0; r < 4; ++r)
            W1[((w + 4 * j) * 16 + lk + 4 * r) * CWP + li] = acc[j][r];
    }
    __syncthreads();

    // ring[h][c] -= sum_g VTt[g][h] * W1[g][c]; same interleaved structure
    {
      const int nfr = R / 16;  // <= 12 (b <= 64)
      v4d acc[3] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};
      const int nj = (nfr - w + 3) / 4;  // frags this wave owns (<= 3)
#pragma unroll 4
      for (int g0 = 0; g0 < G; g0 += 4) {
        const double bb = W1[(g0 + lk) * CWP + li];
#pragma unroll
        for (int j = 0; j < 3; ++j) {
          if (j < nj) {
            const int f = w + 4 * j;
            const double a = VTk[(int64_t)(g0 + lk) * R + f * 16 + li];
            acc[j] = BtMfma<double>::mma(a, bb, acc[j]);
          }
        }
      }
      for (int j = 0; j < nj; ++j) {
        const int f = w + 4 * j;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int slot = sbase + f * 16 + lk + 4 * r;
          if (slot >= R) slot -= R;
          if (slot >= R) slot -= R;
          ring[slot * CWP + li] -= acc[j][r];
        }
      }
    }
    __syncthreads();

    // rows [gbase, gbase + b) are final -> store (their slots are exactly
    // the ones window k+1's fresh rows reuse)
    move_rows(gbase, sbase, 0, b, true);
    __syncthreads();
  }
  // epilogue: flush the remaining R - b rows of the last frame
  const int64_t gl = base0 + (int64_t)(nwin - 1) * b;
  move_rows(gl, (int)(((int64_t)(nwin - 1) * b) % R), b, R - b, true);
}

// complex c128 ------------------------------------------------------------
// Same structure; interleaved (re, im) in ring/W1; 4 MFMA per product with
// conj on the V (left) operand of W1.

__global__ __launch_bounds__(256) void bt_group_c128(
    double* __restrict__ E, int64_t nE, int64_t npad, const double* __restrict__ V,
    const double* __restrict__ VTt, int64_t base0, int b, int G, int R,
    int nwin) {
  extern __shared__ double S[];  // ring[R][2*CWP] then W1[G][2*CWP]
  double* ring = S;
  double* W1 = S + (int64_t)R * 2 * CWP;

  const int tid = threadIdx.x;
  const int lane = tid & 63, w = tid >> 6;
  const int li = lane & 15, lk = lane >> 4;
  const int64_t col0 = (int64_t)blockIdx.x * CW;

  auto move_rows = [&](int64_t gbase, int sbase, int f0, int cnt, bool store) {
    for (int f = f0 + tid / CW; f < f0 + cnt; f += 256 / CW) {
      const int c = tid % CW;
      int slot = sbase + f;
      if (slot >= R) slot -= R;
      if (slot >= R) slot -= R;
      const int64_t grow = gbase + f;
      if (store) {
        if (grow < npad && col0 + c < nE) {
          E[(grow * nE + col0 + c) * 2] = ring[(slot * CWP + c) * 2];
          E[(grow * nE + col0 + c) * 2 + 1] = ring[(slot * CWP + c) * 2 + 1];
        }
      } else {
        double vr = 0.0, vi = 0.0;
        if (grow < npad && col0 + c < nE) {
          vr = E[(grow * nE + col0 + c) * 2];
          vi = E[(grow * nE + col0 + c) * 2 + 1];
        }
        ring[(slot * CWP + c) * 2] = vr;
        ring[(slot * CWP + c) * 2 + 1] = vi;
      }
    }
  };

  move_rows(base0, 0, 0, R - b, false);
  __syncthreads();

  for (int k = 0; k < nwin; ++k) {
    const int64_t gbase = base0 + (int64_t)k * b;
    const int sbase = (int)(((int64_t)k * b) % R);
    move_rows(gbase, sbase, R - b, b, false);
    __syncthreads();

    const double* Vk = V + (int64_t)k * R * G * 2;
    const double* VTk = VTt + (int64_t)k * G * R * 2;

    // W1 = V^H seg: (ar - i ai)(br + i bi); fragment-interleaved
    {
      constexpr int NF = 2;
      v4d ar_[NF] = {{0, 0, 0, 0}, {0, 0, 0, 0}};
      v4d ai_[NF] = {{0, 0, 0, 0}, {0, 0, 0, 0}};
#pragma unroll 2
      for (int h0 = 0; h0 < R; h0 += 4) {
        const int h = h0 + lk;
        int slot = sbase + h;
        if (slot >= R) slot -= R;
        const double br = ring[(slot * CWP + li) * 2];
        const double bi = ring[(slot * CWP + li) * 2 + 1];
#pragma unroll
        for (int j = 0; j < NF; ++j) {
          const int f = w + 4 * j;
          if (f < G / 16) {
            const int64_t va = ((int64_t)h * G + f * 16 + li) * 2;
            const double vr = Vk[va], vi = Vk[va + 1];
            ar_[j] = BtMfma<double>::mma(vr, br, ar_[j]);
            ai_[j] = BtMfma<double>::mma(vr, bi, ai_[j]);
            ar_[j] = BtMfma<double>::mma(vi, bi, ar_[j]);
            ai_[j] = BtMfma<double>::mma(-vi, br, ai_[j]);
          }
        }
      }
#pragma unroll
      for (int j = 0; j < NF; ++j)
        if (w + 4 * j < G / 16)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            W1[(((w + 4 * j) * 16 + lk + 4 * r) * CWP + li) * 2] = ar_[j][r];
            W1[(((w + 4 * j) * 16 + lk + 4 * r) * CWP + li) * 2 + 1] = ai_[j][r];
          }
    }
    __syncthreads();

    {
      const int nfr = R / 16;
      v4d ar_[3] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};
      v4d ai_[3] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};
      const int nj = (nfr - w + 3) / 4;
#pragma unroll 2
      for (int g0 = 0; g0 < G; g0 += 4) {
        const double br = W1[((g0 + lk) * CWP + li) * 2];
        const double bi = W1[((g0 + lk) * CWP + li) * 2 + 1];
#pragma unroll
        for (int j = 0; j < 3; ++j) {
          if (j < nj) {
            const int f = w + 4 * j;
            const int64_t va = ((int64_t)(g0 + lk) * R + f * 16 + li) * 2;
            const double vr = VTk[va], vi = VTk[va + 1];
            ar_[j] = BtMfma<double>::mma(vr, br, ar_[j]);
            ai_[j] = BtMfma<double>::mma(vr, bi, ai_[j]);
            ar_[j] = BtMfma<double>::mma(-vi, bi, ar_[j]);
            ai_[j] = BtMfma<double>::mma(vi, br, ai_[j]);
          }
        }
      }
      for (int j = 0; j < nj; ++j) {
        const int f = w + 4 * j;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int slot = sbase + f * 16 + lk + 4 * r;
          if (slot >= R) slot -= R;
          if (slot >= R) slot -= R;
          ring[(slot * CWP + li) * 2] -= ar_[j][r];
          ring[(slot * CWP + li) * 2 + 1] -= ai_[j][r];
        }
      }
    }
    __syncthreads();

    move_rows(gbase, sbase, 0, b, true);
    __syncthreads();
  }
  const int64_t gl = base0 + (int64_t)(nwin - 1) * b;
  move_rows(gl, (int)(((int64_t)(nwin - 1) * b) % R), b, R - b, true);
}

}  // namespace

extern "C" {

// returns 0 if the configuration is unsupported (caller falls back)
int bt_apply_group_f64(double* E, int64_t nE, int64_t npad, const double* V,
                       const double* VTt, int64_t base0, int b, int G, int R,
                       int nwin, hipStream_t stream) {
  if (G % 32 || b % 16 || R % 16 || R % b || nwin <= 0) return 0;
  const size_t sh = ((size_t)R + G) * CWP * sizeof(double);
  if (sh > 160 * 1024) return 0;
  const int blocks = (int)((nE + CW - 1) / CW);
  if (sh > 65536)
    (void)hipFuncSetAttribute((const void*)bt_group_f64,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              (int)sh);
  bt_group_f64<<<blocks, 256, sh, stream>>>(E, nE, npad, V, VTt, base0, b, G,
                                            R, nwin);
  return 1;
}

int bt_apply_group_c128(double* E, int64_t nE, int64_t npad, const double* V,
                        const double* VTt, int64_t base0, int b, int G, int R,
                        int nwin, hipStream_t stream) {
  if (G % 32 || b % 16 || R % 16 || R % b || nwin <= 0) return 0;
  const size_t sh = ((size_t)R + G) * 2 * CWP * sizeof(double);
  if (sh > 160 * 1024) return 0;
  const int blocks = (int)((nE + CW - 1) / CW);
  if (sh > 65536)
    (void)hipFuncSetAttribute((const void*)bt_group_c128,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              (int)sh);
  bt_group_c128<<<blocks, 256, sh, stream>>>(E, nE, npad, V, VTt, base0, b, G,
                                             R, nwin);
  return 1;
}

}  // extern "C"
