// Band-to-tridiag back-transform: one kernel launch per sweep GROUP.
//
// The compact-WY window chain of bt_band_to_tridiag (reference
// eigensolver/bt_band_to_tridiag/impl.h:59-1031) is sequential in k (windows
// overlap by G-1 rows) but INDEPENDENT across eigenvector columns. The torch
// formulation paid ~3 GEMM launches per window (~70k launches per solve,
// host-launch bound). Here each 256-thread workgroup owns a CW-column slice
// of E and marches the ENTIRE chain of one group in LDS:
//
//   per window k:  W1 = V_k^H seg              (MFMA, K = R ring rows)
//                  seg -= (V_k T_k) W1         (MFMA, K = G)
//                  store the b finalized rows; load the next b rows
//
// The E slice lives in an LDS ring of R = ceil((G+b-1)/b)*b rows; the b-row
// shift between windows makes slot reuse exact (the rows finalized by
// window k free the slots window k+1's fresh rows need). V (padded to R
// rows) and VT^T = (V T)^T (padded to R cols) stream from L2/L3 — shared by
// every workgroup and read once per row-fragment per k-step, so CW sets the
// arithmetic intensity: CW=16 measured V-bandwidth-bound (~78 us/window);
// CW=64 (f64) / 32 (c128) uses the full 160 KiB LDS at one workgroup per CU
// and is MFMA-bound. G and R are template constants (a runtime R defeated
// unrolling: 5 MFMA + AGPR churn in the whole body). Requirements (else the
// torch path runs): G == 128, b % 16 == 0, b <= 64.
#include <hip/hip_runtime.h>

#include <cstdint>

#include "cplx.h"

namespace {

typedef double v4d __attribute__((ext_vector_type(4)));

__device__ inline v4d mma_f64(double a, double b, v4d c) {
  return __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, c, 0, 0, 0);
}

// real f64: CW = 64, LDS = (R + G) * 64 * 8 = 160 KiB at R=192 ------------

template <int G, int R, int CW = 64>
__global__ __launch_bounds__(512, 1) void bt_group_f64(
    double* __restrict__ E, int64_t nE, int64_t ldE, int64_t npad,
    const double* __restrict__ V, const double* __restrict__ VTt,
    int64_t base0, int b, int nwin) {
  constexpr int NW = 8;                     // waves (512 threads)
  constexpr int NRF1 = (G / 16 + NW - 1) / NW;   // W1 row-frags/wave (1)
  constexpr int NRF2 = (R / 16 + NW - 1) / NW;   // update row-frags/wave (2)
  extern __shared__ double S[];  // ring[R][CW] then W1[G][CW]
  double* ring = S;
  double* W1 = S + (int64_t)R * CW;

  const int tid = threadIdx.x;
  const int lane = tid & 63, w = tid >> 6;
  const int li = lane & 15, lk = lane >> 4;
  const int64_t col0 = (int64_t)blockIdx.x * CW;

  auto move_rows = [&](int64_t gbase, int sbase, int f0, int cnt, bool store) {
    for (int f = f0 + tid / CW; f < f0 + cnt; f += (NW * 64) / CW) {
      const int c = tid % CW;
      int slot = sbase + f;
      if (slot >= R) slot -= R;
      if (slot >= R) slot -= R;
      const int64_t grow = gbase + f;
      if (store) {
        if (grow < npad && col0 + c < nE)
          E[grow * ldE + col0 + c] = ring[slot * CW + c];
      } else {
        double v = 0.0;
        if (grow < npad && col0 + c < nE) v = E[grow * ldE + col0 + c];
        ring[slot * CW + c] = v;
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

    const double* Vk = V + (int64_t)k * R * G;
    const double* VTk = VTt + (int64_t)k * G * R;

    // W1[g][c] = sum_h V[h][g] * ring[h][c]; each wave: NRF1 row-frags x 4
    // col-frags; ONE V load per row-frag per k-step shared by col-frags
    {
      v4d acc[NRF1][CW / 16];
#pragma unroll
      for (int j = 0; j < NRF1; ++j)
#pragma unroll
        for (int cf = 0; cf < CW / 16; ++cf) acc[j][cf] = {0, 0, 0, 0};
#pragma unroll 4
      for (int h0 = 0; h0 < R; h0 += 4) {
        const int h = h0 + lk;
        int slot = sbase + h;
        if (slot >= R) slot -= R;
        double a[NRF1];
#pragma unroll
        for (int j = 0; j < NRF1; ++j)
          a[j] = Vk[(int64_t)h * G + (w + NW * j) * 16 + li];
#pragma unroll
        for (int cf = 0; cf < CW / 16; ++cf) {
          const double bb = ring[slot * CW + cf * 16 + li];
#pragma unroll
          for (int j = 0; j < NRF1; ++j)
            acc[j][cf] = mma_f64(a[j], bb, acc[j][cf]);
        }
      }
#pragma unroll
      for (int j = 0; j < NRF1; ++j)
#pragma unroll
        for (int cf = 0; cf < CW / 16; ++cf)
#pragma unroll
          for (int r = 0; r < 4; ++r)
            W1[((w + NW * j) * 16 + lk + 4 * r) * CW + cf * 16 + li] =
                acc[j][cf][r];
    }
    __syncthreads();

    // ring[h][c] -= sum_g VTt[g][h] * W1[g][c]
    {
      const int nfr = R / 16;
      const int nj = (nfr - w + NW - 1) / NW;
      v4d acc[NRF2][CW / 16];
#pragma unroll
      for (int j = 0; j < NRF2; ++j)
#pragma unroll
        for (int cf = 0; cf < CW / 16; ++cf) acc[j][cf] = {0, 0, 0, 0};
#pragma unroll 4
      for (int g0 = 0; g0 < G; g0 += 4) {
        double a[NRF2];
#pragma unroll
        for (int j = 0; j < NRF2; ++j)
          a[j] = (j < nj)
                     ? VTk[(int64_t)(g0 + lk) * R + (w + NW * j) * 16 + li]
                     : 0.0;
#pragma unroll
        for (int cf = 0; cf < CW / 16; ++cf) {
          const double bb = W1[(g0 + lk) * CW + cf * 16 + li];
#pragma unroll
          for (int j = 0; j < NRF2; ++j)
            acc[j][cf] = mma_f64(a[j], bb, acc[j][cf]);
        }
      }
      for (int j = 0; j < nj; ++j) {
        const int f = w + NW * j;
#pragma unroll
        for (int cf = 0; cf < CW / 16; ++cf)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            int slot = sbase + f * 16 + lk + 4 * r;
            if (slot >= R) slot -= R;
            if (slot >= R) slot -= R;
            ring[slot * CW + cf * 16 + li] -= acc[j][cf][r];
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

// complex c128: CW = 32, LDS = (R + G) * 32 * 16 = 160 KiB at R=192 -------

template <int G, int R>
__global__ __launch_bounds__(512, 1) void bt_group_c128(
    double* __restrict__ E, int64_t nE, int64_t npad,
    const double* __restrict__ V, const double* __restrict__ VTt,
    int64_t base0, int b, int nwin) {
  constexpr int CW = 32;
  constexpr int NW = 8;                          // waves (512 threads)
  constexpr int NRF1 = (G / 16 + NW - 1) / NW;   // 1
  constexpr int NRF2 = (R / 16 + NW - 1) / NW;   // <=2
  extern __shared__ double S[];  // ring[R][CW](re,im) then W1[G][CW]
  double* ring = S;
  double* W1 = S + (int64_t)R * CW * 2;

  const int tid = threadIdx.x;
  const int lane = tid & 63, w = tid >> 6;
  const int li = lane & 15, lk = lane >> 4;
  const int64_t col0 = (int64_t)blockIdx.x * CW;

  auto move_rows = [&](int64_t gbase, int sbase, int f0, int cnt, bool store) {
    for (int f = f0 + tid / CW; f < f0 + cnt; f += (NW * 64) / CW) {
      const int c = tid % CW;
      int slot = sbase + f;
      if (slot >= R) slot -= R;
      if (slot >= R) slot -= R;
      const int64_t grow = gbase + f;
      if (store) {
        if (grow < npad && col0 + c < nE) {
          E[(grow * nE + col0 + c) * 2] = ring[(slot * CW + c) * 2];
          E[(grow * nE + col0 + c) * 2 + 1] = ring[(slot * CW + c) * 2 + 1];
        }
      } else {
        double vr = 0.0, vi = 0.0;
        if (grow < npad && col0 + c < nE) {
          vr = E[(grow * nE + col0 + c) * 2];
          vi = E[(grow * nE + col0 + c) * 2 + 1];
        }
        ring[(slot * CW + c) * 2] = vr;
        ring[(slot * CW + c) * 2 + 1] = vi;
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

    // W1 = V^H seg (conj on V); 2 col-frags per wave
    {
      v4d ar[NRF1][2], ai[NRF1][2];
#pragma unroll
      for (int j = 0; j < NRF1; ++j)
#pragma unroll
        for (int cf = 0; cf < 2; ++cf) {
          ar[j][cf] = {0, 0, 0, 0};
          ai[j][cf] = {0, 0, 0, 0};
        }
#pragma unroll 4
      for (int h0 = 0; h0 < R; h0 += 4) {
        const int h = h0 + lk;
        int slot = sbase + h;
        if (slot >= R) slot -= R;
        double avr[NRF1], avi[NRF1];
#pragma unroll
        for (int j = 0; j < NRF1; ++j) {
          const int64_t va = ((int64_t)h * G + (w + NW * j) * 16 + li) * 2;
          avr[j] = Vk[va];
          avi[j] = Vk[va + 1];
        }
#pragma unroll
        for (int cf = 0; cf < 2; ++cf) {
          const double br = ring[(slot * CW + cf * 16 + li) * 2];
          const double bi = ring[(slot * CW + cf * 16 + li) * 2 + 1];
#pragma unroll
          for (int j = 0; j < NRF1; ++j) {
            ar[j][cf] = mma_f64(avr[j], br, ar[j][cf]);
            ai[j][cf] = mma_f64(avr[j], bi, ai[j][cf]);
            ar[j][cf] = mma_f64(avi[j], bi, ar[j][cf]);
            ai[j][cf] = mma_f64(-avi[j], br, ai[j][cf]);
          }
        }
      }
#pragma unroll
      for (int j = 0; j < NRF1; ++j)
#pragma unroll
        for (int cf = 0; cf < 2; ++cf)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int64_t o =
                (((w + NW * j) * 16 + lk + 4 * r) * CW + cf * 16 + li) * 2;
            W1[o] = ar[j][cf][r];
            W1[o + 1] = ai[j][cf][r];
          }
    }
    __syncthreads();

    {
      const int nfr = R / 16;
      const int nj = (nfr - w + NW - 1) / NW;
      v4d ar[NRF2][2], ai[NRF2][2];
#pragma unroll
      for (int j = 0; j < NRF2; ++j)
#pragma unroll
        for (int cf = 0; cf < 2; ++cf) {
          ar[j][cf] = {0, 0, 0, 0};
          ai[j][cf] = {0, 0, 0, 0};
        }
#pragma unroll 4
      for (int g0 = 0; g0 < G; g0 += 4) {
        double avr[NRF2], avi[NRF2];
#pragma unroll
        for (int j = 0; j < NRF2; ++j) {
          if (j < nj) {
            const int64_t va =
                ((int64_t)(g0 + lk) * R + (w + NW * j) * 16 + li) * 2;
            avr[j] = VTk[va];
            avi[j] = VTk[va + 1];
          } else {
            avr[j] = 0.0;
            avi[j] = 0.0;
          }
        }
#pragma unroll
        for (int cf = 0; cf < 2; ++cf) {
          const double br = W1[((g0 + lk) * CW + cf * 16 + li) * 2];
          const double bi = W1[((g0 + lk) * CW + cf * 16 + li) * 2 + 1];
#pragma unroll
          for (int j = 0; j < NRF2; ++j) {
            ar[j][cf] = mma_f64(avr[j], br, ar[j][cf]);
            ai[j][cf] = mma_f64(avr[j], bi, ai[j][cf]);
            ar[j][cf] = mma_f64(-avi[j], bi, ar[j][cf]);
            ai[j][cf] = mma_f64(avi[j], br, ai[j][cf]);
          }
        }
      }
      for (int j = 0; j < nj; ++j) {
        const int f = w + NW * j;
#pragma unroll
        for (int cf = 0; cf < 2; ++cf)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            int slot = sbase + f * 16 + lk + 4 * r;
            if (slot >= R) slot -= R;
            if (slot >= R) slot -= R;
            const int64_t o = (slot * CW + cf * 16 + li) * 2;
            ring[o] -= ar[j][cf][r];
            ring[o + 1] -= ai[j][cf][r];
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

int bt_apply_group_f64(double* E, int64_t nE, int64_t npad, const double* V,
                       const double* VTt, int64_t base0, int b, int G, int R,
                       int nwin, hipStream_t stream) {
  if (G != 128 || b % 16 || b > 64 || R % 16 || R % b || nwin <= 0) return 0;
  const size_t sh64 = ((size_t)R + G) * 64 * sizeof(double);
  if (sh64 > 160 * 1024) return 0;
  // Tail split: Q = ceil(nE/64) column chunks; the last partial CU pass
  // would idle most of the chip, so the remainder columns run as a second
  // CW=32 launch (2 WGs/CU, half the per-WG work) — measured: 313 chunks
  // as 2 full CW64 passes lost to the GEMM chain; 256 + 114x32 wins.
  const int64_t Q = (nE + 63) / 64;
  int64_t main_cols = nE;
  if (Q > 256 && (Q % 256) != 0) main_cols = (Q / 256) * 256 * 64;
  const int64_t rest_cols = nE - main_cols;
  const size_t sh32 = ((size_t)R + G) * 32 * sizeof(double);
#define BT_CASE(RT)                                                        \
  if (R == RT) {                                                           \
    if (main_cols > 0) {                                                   \
      const void* fp = (const void*)bt_group_f64<128, RT, 64>;             \
      if (sh64 > 65536)                                                    \
        (void)hipFuncSetAttribute(                                         \
            fp, hipFuncAttributeMaxDynamicSharedMemorySize, (int)sh64);    \
      bt_group_f64<128, RT, 64>                                            \
          <<<(int)((main_cols + 63) / 64), 512, sh64, stream>>>(           \
              E, main_cols, nE, npad, V, VTt, base0, b, nwin);             \
    }                                                                      \
    if (rest_cols > 0) {                                                   \
      const void* fp = (const void*)bt_group_f64<128, RT, 32>;             \
      if (sh32 > 65536)                                                    \
        (void)hipFuncSetAttribute(                                         \
            fp, hipFuncAttributeMaxDynamicSharedMemorySize, (int)sh32);    \
      bt_group_f64<128, RT, 32>                                            \
          <<<(int)((rest_cols + 31) / 32), 512, sh32, stream>>>(           \
              E + main_cols, rest_cols, nE, npad, V, VTt, base0, b, nwin); \
    }                                                                      \
    return 1;                                                              \
  }
  BT_CASE(192) BT_CASE(176) BT_CASE(160) BT_CASE(144)
#undef BT_CASE
  return 0;
}

int bt_apply_group_c128(double* E, int64_t nE, int64_t npad, const double* V,
                        const double* VTt, int64_t base0, int b, int G, int R,
                        int nwin, hipStream_t stream) {
  if (G != 128 || b % 16 || b > 64 || R % 16 || R % b || nwin <= 0) return 0;
  const size_t sh = ((size_t)R + G) * 32 * 2 * sizeof(double);
  if (sh > 160 * 1024) return 0;
  const int blocks = (int)((nE + 31) / 32);
#define BT_CASE(RT)                                                        \
  if (R == RT) {                                                           \
    const void* fp = (const void*)bt_group_c128<128, RT>;                  \
    if (sh > 65536)                                                        \
      (void)hipFuncSetAttribute(fp,                                        \
                                hipFuncAttributeMaxDynamicSharedMemorySize,\
                                (int)sh);                                  \
    bt_group_c128<128, RT><<<blocks, 512, sh, stream>>>(E, nE, npad, V,    \
                                                        VTt, base0, b,     \
                                                        nwin);             \
    return 1;                                                              \
  }
  BT_CASE(192) BT_CASE(176) BT_CASE(160) BT_CASE(144)
#undef BT_CASE
  return 0;
}

}  // extern "C"
