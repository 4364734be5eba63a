// Single-tile factorization building blocks for CDNA4.
//
// * potrf_block: single-workgroup LDS-resident Cholesky of a leading n x n
//   block (n <= 128 real / n <= 64 complex, limited by 160 KiB LDS). The tile
//   Cholesky of an nb x nb tile is orchestrated on the host (ext.cpp) as
//   potrf_block + column-parallel block inverse + fused GEMM panel/trailing
//   updates — the per-tile analog of the right-looking algorithm, replacing the
//   reference's rocSOLVER potrf call.
// * trtri_lower: column-parallel lower-triangular inverse (thread-per-column
//   forward substitution). Triangular solves everywhere in the library are
//   GEMMs against these block inverses — the standard GPU TRSM trade.
#include "kernels.h"
#include "cplx.h"

namespace {

template <typename S, int BSZ>
__launch_bounds__(256) __global__ void potrf_block_k(S* A, int n, int ld) {
  using TR = ScalarTraits<S>;
  using RT = typename TR::real_t;
  __shared__ S L[BSZ][BSZ + 1];
  const int tid = threadIdx.x;

  // load (full block; only the lower triangle is referenced)
  for (int e = tid; e < BSZ * BSZ; e += 256) {
    const int i = e / BSZ, j = e % BSZ;
    L[i][j] = (i < n && j < n) ? A[(int64_t)i * ld + j] : TR::zero();
  }
  __syncthreads();

  const int ncb = (n + 15) / 16;
  for (int cb = 0; cb < ncb; ++cb) {
    const int c0 = cb * 16;
    const int bsz = min(16, n - c0);
    // 1) factor the 16x16 diagonal block with DEFERRED column scaling
    //    (LDL^T-style elimination on unscaled columns, one barrier per step:
    //     U[i][j] -= U[i][p] conj(U[j][p]) / U[p][p], then L[:,j] = U[:,j] /
    //     sqrt(U[j][j]) in one scale pass).
    {
      const int i = tid / 16, j = tid % 16;
      for (int p = 0; p < bsz; ++p) {
        const RT s = RT(1) / TR::real(L[c0 + p][c0 + p]);
        if (i < bsz && j < bsz && j > p && j <= i)
          L[c0 + i][c0 + j] -= (L[c0 + i][c0 + p] * TR::conj(L[c0 + j][c0 + p])) * s;
        __syncthreads();
      }
      // snapshot the column scale BEFORE any thread rewrites the diagonal
      const bool act = (i < bsz && j < bsz && j <= i);
      const RT sc = act ? RT(1) / sqrt(TR::real(L[c0 + j][c0 + j])) : RT(1);
      __syncthreads();
      if (act) L[c0 + i][c0 + j] = L[c0 + i][c0 + j] * sc;
      __syncthreads();
    }
    // 2) panel solve: rows below vs the diagonal block (X * D^H = A)
    {
      const int r = c0 + 16 + tid;
      if (r < n) {
        for (int j = 0; j < bsz; ++j) {
          S x = L[r][c0 + j];
          for (int p = 0; p < j; ++p) x -= L[r][c0 + p] * TR::conj(L[c0 + j][c0 + p]);
          L[r][c0 + j] = x * (RT(1) / TR::real(L[c0 + j][c0 + j]));
        }
      }
      __syncthreads();
    }
    // 3) trailing update (lower part only)
    {
      const int t0 = c0 + 16;
      const int nt = n - t0;
      if (nt > 0) {
        for (int e = tid; e < nt * nt; e += 256) {
          const int i = e / nt, j = e % nt;
          if (j <= i) {
            S s = L[t0 + i][c0];
            S acc = s * TR::conj(L[t0 + j][c0]);
            for (int p = 1; p < bsz; ++p)
              acc += L[t0 + i][c0 + p] * TR::conj(L[t0 + j][c0 + p]);
            L[t0 + i][t0 + j] -= acc;
          }
        }
      }
      __syncthreads();
    }
  }

  // store lower triangle (incl. diagonal)
  for (int e = tid; e < BSZ * BSZ; e += 256) {
    const int i = e / BSZ, j = e % BSZ;
    if (i < n && j < n && j <= i) A[(int64_t)i * ld + j] = L[i][j];
  }
}

// Single-workgroup LDS-resident variant for blocks (n <= 128 real / 64
// complex): staging through LDS removes the global-latency-bound serial chain
// of the naive version (measured 21.8 ms -> tens of us for a 512 tile's
// blocks). One thread per column, forward substitution in LDS.
template <typename S, int BSZ>
__launch_bounds__(256) __global__ void trtri_block_lds_k(const S* L, S* T,
                                                         int n, int ldl,
                                                         int ldt,
                                                         int unit_diag) {
  using TR = ScalarTraits<S>;
  __shared__ S Ts[BSZ][BSZ + 1];  // the own-write substitution chain (latency-critical)
  const int tid = threadIdx.x;
  const int j = tid;
  if (j < n) {
    for (int i = 0; i < j; ++i) Ts[i][j] = TR::zero();
    const S djj = unit_diag ? TR::from_real(1) : TR::recip(L[(int64_t)j * ldl + j]);
    Ts[j][j] = djj;
    for (int i = j + 1; i < n; ++i) {
      S acc = TR::zero();
      const S* Lrow = L + (int64_t)i * ldl;  // L2-shared across all columns
      for (int p = j; p < i; ++p) acc += Lrow[p] * Ts[p][j];
      const S dii = unit_diag ? TR::from_real(1) : TR::recip(L[(int64_t)i * ldl + i]);
      Ts[i][j] = -(dii * acc);
    }
  }
  __syncthreads();
  for (int e = tid; e < BSZ * BSZ; e += 256) {
    const int i = e / BSZ, j2 = e % BSZ;
    if (i < n && j2 < n) T[(int64_t)i * ldt + j2] = Ts[i][j2];
  }
}

// Fused factor + invert of one diagonal block, single workgroup.
//
// Real BSZ=128: LDS = L[128][128] (131 KiB) + S scratch 64x64 (32 KiB) = the
// full 160 KiB CU budget. The inverse T = L^-1 is computed blockwise with a
// 64-split, recycling freed LDS regions of L:
//   T11 = inv(L11) -> S -> (global, and copied over L11's region)
//   W   = L21*T11  -> S   (L21 still live)
//   T22 = inv(L22) -> L21's region -> global  (L21 dead after W)
//   T21 = -T22*W   -> global
// Complex BSZ=64: direct column-parallel inversion into S (both fit).
// The dinv output block is always BSZ x BSZ, identity-extended when n < BSZ.
//
// do_factor=0 skips the factorization phases (inverting an already-factored
// received tile after a broadcast).
template <typename S, int BSZ>
__launch_bounds__(256) __global__ void potrf_invert_block_k(S* A, int n,
                                                            int ld, S* Tout,
                                                            int do_factor) {
  using TR = ScalarTraits<S>;
  using RT = typename TR::real_t;
  constexpr int HB = BSZ / 2;
  __shared__ S L[BSZ][BSZ];
  __shared__ S Sc[HB][HB];  // no pad: L + Sc is exactly the 160 KiB CU budget
  const int tid = threadIdx.x;

  for (int e = tid; e < BSZ * BSZ; e += 256) {
    const int i = e / BSZ, j = e % BSZ;
    L[i][j] = (i < n && j < n) ? A[(int64_t)i * ld + j] : TR::zero();
  }
  __syncthreads();

  if (do_factor) {
    const int ncb = (n + 15) / 16;
    const int ti = tid / 16, tj = tid % 16;
    for (int cb = 0; cb < ncb; ++cb) {
      const int c0 = cb * 16;
      const int bsz16 = min(16, n - c0);
      // deferred-scaling 16x16 diagonal factor (see potrf_block_k)
      for (int p = 0; p < bsz16; ++p) {
        const RT s = RT(1) / TR::real(L[c0 + p][c0 + p]);
        if (ti < bsz16 && tj < bsz16 && tj > p && tj <= ti)
          L[c0 + ti][c0 + tj] -= (L[c0 + ti][c0 + p] * TR::conj(L[c0 + tj][c0 + p])) * s;
        __syncthreads();
      }
      const bool act = (ti < bsz16 && tj < bsz16 && tj <= ti);
      const RT sc = act ? RT(1) / sqrt(TR::real(L[c0 + tj][c0 + tj])) : RT(1);
      __syncthreads();
      if (act) L[c0 + ti][c0 + tj] = L[c0 + ti][c0 + tj] * sc;
      __syncthreads();
      // panel solve below the 16-block
      {
        const int r = c0 + 16 + tid;
        if (r < n) {
          for (int j = 0; j < bsz16; ++j) {
            S x = L[r][c0 + j];
            for (int p = 0; p < j; ++p) x -= L[r][c0 + p] * TR::conj(L[c0 + j][c0 + p]);
            L[r][c0 + j] = x * (RT(1) / TR::real(L[c0 + j][c0 + j]));
          }
        }
        __syncthreads();
      }
      // trailing update (lower)
      {
        const int t0 = c0 + 16;
        const int nt = n - t0;
        if (nt > 0) {
          for (int e = tid; e < nt * nt; e += 256) {
            const int i = e / nt, j = e % nt;
            if (j <= i) {
              S acc = L[t0 + i][c0] * TR::conj(L[t0 + j][c0]);
              for (int p = 1; p < bsz16; ++p)
                acc += L[t0 + i][c0 + p] * TR::conj(L[t0 + j][c0 + p]);
              L[t0 + i][t0 + j] -= acc;
            }
          }
        }
        __syncthreads();
      }
    }
    // store the factor (lower triangle)
    for (int e = tid; e < BSZ * BSZ; e += 256) {
      const int i = e / BSZ, j = e % BSZ;
      if (i < n && j < n && j <= i) A[(int64_t)i * ld + j] = L[i][j];
    }
    // no barrier needed: inversion only reads LDS L, which is final
  }

  // ---- inversion ----
  // initialize Tout = identity-extended zero (overwritten below where computed)
  for (int e = tid; e < BSZ * BSZ; e += 256) {
    const int i = e / BSZ, j = e % BSZ;
    Tout[i * BSZ + j] = (i == j && i >= n) ? TR::from_real(1) : TR::zero();
  }
  __syncthreads();  // order init writes before the computed overwrites

  if constexpr (BSZ == 64) {
    // blocked-16 inversion: serial chains shrink from 64 to 16 steps (the
    // direct column-parallel form was a ~2000-FMA dependent chain per thread
    // and dominated the 108us kernel time); off-diagonal 16x16 blocks are
    // thread-parallel products  T_IJ = -T_II * (sum_P L_IP T_PJ).
    __shared__ S Tc[64][65];
    for (int e = tid; e < 64 * 65; e += 256) Tc[e / 65][e % 65] = TR::zero();
    __syncthreads();
    // diagonal 16-blocks, one column per thread (64 threads, 16-step chains)
    {
      const int j = tid;
      if (j < n) {
        const int B0 = (j / 16) * 16;
        const int bend = min(B0 + 16, n);
        Tc[j][j] = TR::recip(L[j][j]);
        for (int i = j + 1; i < bend; ++i) {
          S acc = TR::zero();
          for (int p = j; p < i; ++p) acc += L[i][p] * Tc[p][j];
          Tc[i][j] = -(TR::recip(L[i][i]) * acc);
        }
      }
    }
    __syncthreads();
    // off-diagonal blocks by distance d: W = sum_P L_IP T_PJ, T_IJ = -T_II W
    for (int dist = 1; dist < 4; ++dist) {
      const int nblk = 4 - dist;
      // phase 1: W into Sc (block b at rows (b/2)*16, cols (b%2)*16)
      for (int e = tid; e < nblk * 256; e += 256) {
        const int b = e / 256, r = (e % 256) / 16, c = e % 16;
        const int I = (b + dist) * 16, J = b * 16;
        S acc = TR::zero();
        for (int p = J; p < I; ++p) acc += L[I + r][p] * Tc[p][J + c];
        Sc[(b / 2) * 16 + r][(b % 2) * 16 + c] = acc;
      }
      __syncthreads();
      // phase 2: T_IJ = -T_II * W (T_II lower triangular)
      for (int e = tid; e < nblk * 256; e += 256) {
        const int b = e / 256, r = (e % 256) / 16, c = e % 16;
        const int I = (b + dist) * 16, J = b * 16;
        S acc = TR::zero();
        for (int p = 0; p <= r; ++p)
          acc += Tc[I + r][I + p] * Sc[(b / 2) * 16 + p][(b % 2) * 16 + c];
        Tc[I + r][J + c] = -acc;
      }
      __syncthreads();
    }
    for (int e = tid; e < BSZ * BSZ; e += 256) {
      const int i = e / BSZ, j2 = e % BSZ;
      if (i < n && j2 < n) Tout[i * BSZ + j2] = Tc[i][j2];
    }
    return;
  }

  // BSZ == 128: 64-split scheme
  const int h = min(HB, n);
  const int rest = n - h;
  // T11 = inv(L11) -> Sc
  {
    const int j = tid;
    if (j < h) {
      for (int i = 0; i < j; ++i) Sc[i][j] = TR::zero();
      Sc[j][j] = TR::recip(L[j][j]);
      for (int i = j + 1; i < h; ++i) {
        S acc = TR::zero();
        for (int p = j; p < i; ++p) acc += L[i][p] * Sc[p][j];
        Sc[i][j] = -(TR::recip(L[i][i]) * acc);
      }
    }
    __syncthreads();
    for (int e = tid; e < h * h; e += 256) {
      const int i = e / h, j2 = e % h;
      Tout[i * BSZ + j2] = Sc[i][j2];
    }
  }
  if (rest > 0) {
    // W = L21 * T11 -> needs T11 (Sc) and L21 (L); write W over L11's region
    // (L11 is dead). Then T22 = inv(L22) into Sc (Sc free after W copy? no —
    // W lives in L11's region, Sc holds T11 still needed? T21 = -T22*W only
    // needs W and T22. So: W -> L11 region, T22 -> Sc (overwrite T11).
    __syncthreads();
    for (int e = tid; e < rest * h; e += 256) {
      const int i = e / h, j = e % h;
      S acc = TR::zero();
      for (int p = j; p < h; ++p) acc += L[h + i][p] * Sc[p][j];
      L[i][j] = acc;  // W[i][j] stored in L11's region (row i < 64)
    }
    __syncthreads();
    // T22 = inv(L22) -> Sc
    {
      const int j = tid;
      if (j < rest) {
        for (int i = 0; i < j; ++i) Sc[i][j] = TR::zero();
        Sc[j][j] = TR::recip(L[h + j][h + j]);
        for (int i = j + 1; i < rest; ++i) {
          S acc = TR::zero();
          for (int p = j; p < i; ++p) acc += L[h + i][h + p] * Sc[p][j];
          Sc[i][j] = -(TR::recip(L[h + i][h + i]) * acc);
        }
      }
      __syncthreads();
      for (int e = tid; e < rest * rest; e += 256) {
        const int i = e / rest, j2 = e % rest;
        Tout[(h + i) * BSZ + (h + j2)] = Sc[i][j2];
      }
    }
    // T21 = -T22 * W  (T22 in Sc, W in L11 region)
    for (int e = tid; e < rest * h; e += 256) {
      const int i = e / h, j = e % h;
      S acc = TR::zero();
      for (int p = 0; p <= i && p < rest; ++p) acc += Sc[i][p] * L[p][j];
      Tout[(h + i) * BSZ + j] = -acc;
    }
  }
}

template <typename S>
__global__ void trtri_lower_k(const S* L, S* T, int n, int ldl, int ldt,
                              int unit_diag) {
  using TR = ScalarTraits<S>;
  const int j = blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= n) return;
  // zero the upper part of column j (full tiles feed fused GEMMs)
  for (int i = 0; i < j; ++i) T[(int64_t)i * ldt + j] = TR::zero();
  const S djj = unit_diag ? TR::from_real(1) : TR::recip(L[(int64_t)j * ldl + j]);
  T[(int64_t)j * ldt + j] = djj;
  for (int i = j + 1; i < n; ++i) {
    S acc = TR::zero();
    for (int p = j; p < i; ++p)
      acc += L[(int64_t)i * ldl + p] * T[(int64_t)p * ldt + j];
    const S dii = unit_diag ? TR::from_real(1) : TR::recip(L[(int64_t)i * ldl + i]);
    T[(int64_t)i * ldt + j] = -(dii * acc);
  }
}

}  // namespace

extern "C" {

// Fused factor (optional) + block inverse. Tout is the BSZ x BSZ dinv block.
// BSZ = 64 for ALL dtypes: the 64-block kernel needs ~73 KiB LDS, so it can
// co-reside with trailing-update GEMM blocks on a CU — a 128-block (160 KiB)
// kernel can never be scheduled next to them and serializes the lookahead.
void potrf_invert_block_f64(double* A, int n, int ld, double* Tout,
                            int do_factor, hipStream_t stream) {
  potrf_invert_block_k<double, 64>
      <<<1, 256, 0, stream>>>(A, n, ld, Tout, do_factor);
}
void potrf_invert_block_f32(float* A, int n, int ld, float* Tout,
                            int do_factor, hipStream_t stream) {
  potrf_invert_block_k<float, 64>
      <<<1, 256, 0, stream>>>(A, n, ld, Tout, do_factor);
}
void potrf_invert_block_c128(double* A, int n, int ld, double* Tout,
                             int do_factor, hipStream_t stream) {
  potrf_invert_block_k<cplx<double>, 64><<<1, 256, 0, stream>>>(
      reinterpret_cast<cplx<double>*>(A), n, ld,
      reinterpret_cast<cplx<double>*>(Tout), do_factor);
}
void potrf_invert_block_c64(float* A, int n, int ld, float* Tout,
                            int do_factor, hipStream_t stream) {
  potrf_invert_block_k<cplx<float>, 64><<<1, 256, 0, stream>>>(
      reinterpret_cast<cplx<float>*>(A), n, ld,
      reinterpret_cast<cplx<float>*>(Tout), do_factor);
}

void potrf_block128_f64(double* A, int n, int ld, hipStream_t stream) {
  potrf_block_k<double, 128><<<1, 256, 0, stream>>>(A, n, ld);
}
void potrf_block128_f32(float* A, int n, int ld, hipStream_t stream) {
  potrf_block_k<float, 128><<<1, 256, 0, stream>>>(A, n, ld);
}
// complex: block size 64 (LDS limit); `A` is the interleaved base, ld in
// complex elements.
void potrf_block128_c128(double* A, int n, int ld, hipStream_t stream) {
  potrf_block_k<cplx<double>, 64>
      <<<1, 256, 0, stream>>>(reinterpret_cast<cplx<double>*>(A), n, ld);
}
void potrf_block128_c64(float* A, int n, int ld, hipStream_t stream) {
  potrf_block_k<cplx<float>, 64>
      <<<1, 256, 0, stream>>>(reinterpret_cast<cplx<float>*>(A), n, ld);
}

void trtri_lower_f64(const double* L, double* T, int n, int ldl, int ldt,
                     int unit_diag, hipStream_t stream) {
  if (n <= 128)
    trtri_block_lds_k<double, 128>
        <<<1, 256, 0, stream>>>(L, T, n, ldl, ldt, unit_diag);
  else
    trtri_lower_k<double>
        <<<(n + 255) / 256, 256, 0, stream>>>(L, T, n, ldl, ldt, unit_diag);
}
void trtri_lower_f32(const float* L, float* T, int n, int ldl, int ldt,
                     int unit_diag, hipStream_t stream) {
  if (n <= 128)
    trtri_block_lds_k<float, 128>
        <<<1, 256, 0, stream>>>(L, T, n, ldl, ldt, unit_diag);
  else
    trtri_lower_k<float>
        <<<(n + 255) / 256, 256, 0, stream>>>(L, T, n, ldl, ldt, unit_diag);
}
void trtri_lower_c128(const double* L, double* T, int n, int ldl, int ldt,
                      int unit_diag, hipStream_t stream) {
  auto Lc = reinterpret_cast<const cplx<double>*>(L);
  auto Tc = reinterpret_cast<cplx<double>*>(T);
  if (n <= 64)
    trtri_block_lds_k<cplx<double>, 64>
        <<<1, 256, 0, stream>>>(Lc, Tc, n, ldl, ldt, unit_diag);
  else
    trtri_lower_k<cplx<double>>
        <<<(n + 255) / 256, 256, 0, stream>>>(Lc, Tc, n, ldl, ldt, unit_diag);
}
void trtri_lower_c64(const float* L, float* T, int n, int ldl, int ldt,
                     int unit_diag, hipStream_t stream) {
  auto Lc = reinterpret_cast<const cplx<float>*>(L);
  auto Tc = reinterpret_cast<cplx<float>*>(T);
  if (n <= 128)
    trtri_block_lds_k<cplx<float>, 128>
        <<<1, 256, 0, stream>>>(Lc, Tc, n, ldl, ldt, unit_diag);
  else
    trtri_lower_k<cplx<float>>
        <<<(n + 255) / 256, 256, 0, stream>>>(Lc, Tc, n, ldl, ldt, unit_diag);
}

}  // extern "C"
