// Fused batched tile GEMM v2 for CDNA4 (gfx950) — full-tile fast path.
//
// Round-2 rewrite of the v1 kernel driven by the round-1 ISA analysis
// (profiles/gemm_isa_analysis.md): the v1 loop was a serial stage->compute
// structure with 8-byte global loads and ~650 addressing instructions per
// 32 MFMA (MfmaUtil 57.5%). v2 fixes all three findings at once by staging
// through LDS-DMA (`global_load_lds_dwordx4`, 16 B per lane):
//   * no register staging, no ds_write pass, no per-element address VALU —
//     each 1024-B LDS row is ONE instruction with per-lane global addresses
//     held in incremented registers;
//   * ping-pong LDS buffers with ONE barrier per K-step: the DMA of step
//     s+1 is issued before the MFMA block of step s, so its latency lands
//     under the 4096-cycle f64 MFMA span;
//   * 128x128 C blocks (4 waves, 64x64 per wave, 4x4 fragments of
//     mfma_f64_16x16x4) double the flop:LDS-byte ratio of v1's 128x64.
//
// Scope: full tiles only (M%128==0, N%128==0, K%16==0 real; 64/64/16
// complex), not in-place. Edge shapes and in-place panel applies stay on
// the guarded v1 kernel (gemm_tiles.hip). Role parity: the per-tile gemm
// of the reference (/root/reference/include/dlaf/blas/tile.h:352) — here
// one launch covers a whole fused phase.
#include "kernels.h"
#include <type_traits>

typedef double v4d __attribute__((ext_vector_type(4)));
typedef float v4f __attribute__((ext_vector_type(4)));

namespace {

template <typename T>
struct MfmaV2;
template <>
struct MfmaV2<double> {
  using acc_t = v4d;
  static __device__ inline acc_t mma(double a, double b, acc_t c) {
    return __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, c, 0, 0, 0);
  }
  // measured on gfx950 (tools/probe_mfma_f64.hip): D row = lk + 4*reg
  static __device__ inline int acc_row(int lk, int r) { return lk + 4 * r; }
};
template <>
struct MfmaV2<float> {
  using acc_t = v4f;
  static __device__ inline acc_t mma(float a, float b, acc_t c) {
    return __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, c, 0, 0, 0);
  }
  static __device__ inline int acc_row(int lk, int r) { return lk * 4 + r; }
};

__device__ inline void glds16(const void* g, void* lds) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)g,
      (__attribute__((address_space(3))) void*)lds, 16, 0, 0);
}

// Bijective XCD remap (8 XCDs): consecutive output ids land on one XCD so
// the blocks of one desc (which share A/B slabs) co-reside in one L2.
__device__ inline int xcd_remap(int wg, int nwg) {
  const int nx = 8;
  const int q = nwg / nx, r = nwg % nx;
  const int xcd = wg % nx, pos = wg / nx;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
}

// ---------------- real kernel ----------------
// BM=BN=128, BK=16, 256 threads; wave w owns the 64x64 quadrant
// (wrow, wcol) = ((w>>1)*64, (w&1)*64).
// LDS (one __shared__ object — §5 trap 4a): S[buf][A:2048 | B:2048] elems.
//   A image: OPA==N -> row-major [i][k] (glds row 8i-group), else k-major
//   [k][i]; B image: OPB==N -> k-major [k][c], else row-major [c][k].
//   Every image is glds-lane-linear: instr t covers LDS elems t*128..+127.
template <typename T, int OPA, int OPB, int SWZ>
__launch_bounds__(256, 2) __global__ void gemm_v2_k(
    const GemmDesc* __restrict__ descs, const T* __restrict__ A,
    const T* __restrict__ B, T* __restrict__ C, int M, int N, int K, int lda,
    int ldb, int ldc, T alpha, T beta, int mblocks, int nblocks) {
  constexpr int BK = 16;
  constexpr int EPL = 16 / sizeof(T);  // elements per lane per glds
  __shared__ T S[2][4096 * (8 / sizeof(T))];

  int wg = blockIdx.x;
  if constexpr (SWZ) wg = xcd_remap(wg, gridDim.x);
  const int per_desc = mblocks * nblocks;
  const GemmDesc d = descs[wg / per_desc];
  const int rem = wg % per_desc;
  const int i0 = (rem / nblocks) * 128, j0 = (rem % nblocks) * 128;

  const int tid = threadIdx.x;
  const int lane = tid & 63, w = tid >> 6;
  const int wrow = (w >> 1) * 64, wcol = (w & 1) * 64;
  const int li = lane & 15, lk = lane >> 4;

  const int spt = K / BK;  // steps per K-tile
  const int total = (int)d.ktiles * spt;
  // glds instructions per operand image (2048 elems) and per wave
  constexpr int NAW = 2048 / (64 * EPL) / 4;  // f64: 4, f32: 2

  using acc_t = typename MfmaV2<T>::acc_t;
  acc_t acc[4][4];
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int b = 0; b < 4; ++b) acc[a][b] = {0, 0, 0, 0};

  // per-lane glds source offsets for instr j=0 of this wave (element units,
  // relative to the K-tile base) and the per-instr / per-step increments
  const T* Abase0 = A + d.a_off;
  const T* Bbase0 = B + d.b_off;
  int64_t a_l0, a_jinc, a_sinc;
  if constexpr (OPA == OP_N) {  // [i][k] image: one instr = 1024 B of rows
    constexpr int RPI = 64 * EPL / BK;  // rows per instr (f64: 8, f32: 16)
    a_l0 = (int64_t)(i0 + (NAW * w) * RPI + lane / (BK / EPL)) * lda +
           (lane % (BK / EPL)) * EPL;
    a_jinc = (int64_t)RPI * lda;
    a_sinc = BK;
  } else {  // [k][i] image: one instr = whole 128-elem k-rows
    constexpr int KPI = 64 * EPL / 128;  // k-rows per instr (f64: 1, f32: 2)
    a_l0 = (int64_t)(NAW * w * KPI + lane / (128 / EPL)) * lda + i0 +
           (lane % (128 / EPL)) * EPL;
    a_jinc = (int64_t)KPI * lda;
    a_sinc = (int64_t)BK * lda;
  }
  int64_t b_l0, b_jinc, b_sinc;
  if constexpr (OPB == OP_N) {  // [k][c] image
    constexpr int KPI = 64 * EPL / 128;
    b_l0 = (int64_t)(NAW * w * KPI + lane / (128 / EPL)) * ldb + j0 +
           (lane % (128 / EPL)) * EPL;
    b_jinc = (int64_t)KPI * ldb;
    b_sinc = (int64_t)BK * ldb;
  } else {  // [c][k] image
    constexpr int RPI = 64 * EPL / BK;
    b_l0 = (int64_t)(j0 + (NAW * w) * RPI + lane / (BK / EPL)) * ldb +
           (lane % (BK / EPL)) * EPL;
    b_jinc = (int64_t)RPI * ldb;
    b_sinc = BK;
  }

  // running state of the issue pipeline
  const T* aP = Abase0 + a_l0;
  const T* bP = Bbase0 + b_l0;
  int ks_cnt = 0;
  int64_t kt = 0;
  T* const Sflat = &S[0][0];
  constexpr int BUFE = 4096 * (8 / sizeof(T));  // elems per LDS buffer
  constexpr int BOFF = 2048 * (8 / sizeof(T));  // B image offset in a buffer
  const int ldsw = w * NAW * 64 * EPL;  // this wave's LDS elem offset (j=0)

  auto issue = [&](int buf) {
#pragma unroll
    for (int j = 0; j < NAW; ++j)
      glds16(aP + j * a_jinc, Sflat + buf * BUFE + ldsw + j * 64 * EPL);
#pragma unroll
    for (int j = 0; j < NAW; ++j)
      glds16(bP + j * b_jinc, Sflat + buf * BUFE + BOFF + ldsw + j * 64 * EPL);
    // advance to the next K step (next K tile when this one is done)
    if (++ks_cnt == spt) {
      ks_cnt = 0;
      ++kt;
      aP = Abase0 + kt * d.a_kstride + a_l0;
      bP = Bbase0 + kt * d.b_kstride + b_l0;
    } else {
      aP += a_sinc;
      bP += b_sinc;
    }
  };

  auto compute = [&](int buf) {
    const T* sa = Sflat + buf * BUFE;
    const T* sb = sa + BOFF;
    // fragment index helpers for the two image layouts
    auto aidx = [&](int row, int k) {
      return (OPA == OP_N) ? row * BK + k : k * 128 + row;
    };
    auto bidx = [&](int col, int k) {
      return (OPB == OP_N) ? k * 128 + col : col * BK + k;
    };
    T af[2][4], bf[2][4];
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) af[0][mi] = sa[aidx(wrow + mi * 16 + li, lk)];
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) bf[0][ni] = sb[bidx(wcol + ni * 16 + li, lk)];
#pragma unroll
    for (int ks = 0; ks < BK / 4; ++ks) {
      const int cur = ks & 1;
      if (ks + 1 < BK / 4) {
        const int nxt = 1 - cur;
        const int k = (ks + 1) * 4 + lk;
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
          af[nxt][mi] = sa[aidx(wrow + mi * 16 + li, k)];
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          bf[nxt][ni] = sb[bidx(wcol + ni * 16 + li, k)];
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = MfmaV2<T>::mma(af[cur][mi], bf[cur][ni], acc[mi][ni]);
    }
  };

  issue(0);
  for (int s = 0; s < total; ++s) {
    __syncthreads();  // drains the in-flight glds (vmcnt 0) + LDS reuse
    if (s + 1 < total) issue(1 - (s & 1));
    compute(s & 1);
  }

  // epilogue: C = alpha*acc + beta*C (beta test hoisted out of the loops)
  T* Cb = C + d.c_off;
  auto store = [&](auto betanz) {
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const acc_t v = acc[mi][ni];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = i0 + wrow + mi * 16 + MfmaV2<T>::acc_row(lk, r);
          const int col = j0 + wcol + ni * 16 + li;
          const int64_t off = (int64_t)row * ldc + col;
          T out = alpha * (T)v[r];
          if constexpr (decltype(betanz)::value) out += beta * Cb[off];
          Cb[off] = out;
        }
      }
  };
  if (beta != T(0))
    store(std::true_type{});
  else
    store(std::false_type{});
}

// ---------------- complex kernel ----------------
// Interleaved (re,im); one lane's 16-B glds = one complex f64 element (f32:
// two). BM=BN=64, BK=16; wave w owns the 32x32 quadrant; 2x2 fragments,
// 4 MFMA per fragment pair. LDS images mirror the real kernel with complex
// elements; fragments are read as (re,im) pairs in one ds_read_b128 (f64).
template <typename T, int OPA, int OPB, int SWZ>
__launch_bounds__(256, 2) __global__ void gemm_v2_cplx_k(
    const GemmDesc* __restrict__ descs, const T* __restrict__ A,
    const T* __restrict__ B, T* __restrict__ C, int M, int N, int K, int lda,
    int ldb, int ldc, T alpha_re, T alpha_im, T beta_re, T beta_im,
    int mblocks, int nblocks) {
  constexpr int BK = 16;
  constexpr int CPL = 8 / sizeof(T);  // complex elems per lane per glds
  // S[buf][A: 64*16 | B: 16*64] complex elems, interleaved re/im
  __shared__ T S[2][4096 * (8 / sizeof(T))];

  int wg = blockIdx.x;
  if constexpr (SWZ) wg = xcd_remap(wg, gridDim.x);
  const int per_desc = mblocks * nblocks;
  const GemmDesc d = descs[wg / per_desc];
  const int rem = wg % per_desc;
  const int i0 = (rem / nblocks) * 64, j0 = (rem % nblocks) * 64;

  const int tid = threadIdx.x;
  const int lane = tid & 63, w = tid >> 6;
  const int wrow = (w >> 1) * 32, wcol = (w & 1) * 32;
  const int li = lane & 15, lk = lane >> 4;

  const int spt = K / BK;
  const int total = (int)d.ktiles * spt;
  constexpr int NAW = 1024 / (64 * CPL) / 4;  // glds per wave (c128: 4, c64: 2)

  using acc_t = typename MfmaV2<T>::acc_t;
  acc_t accr[2][2], acci[2][2];
#pragma unroll
  for (int a = 0; a < 2; ++a)
#pragma unroll
    for (int b = 0; b < 2; ++b) {
      accr[a][b] = {0, 0, 0, 0};
      acci[a][b] = {0, 0, 0, 0};
    }

  const T* Abase0 = A + 2 * d.a_off;
  const T* Bbase0 = B + 2 * d.b_off;
  // offsets in COMPLEX elements (scaled by 2 at use)
  int64_t a_l0, a_jinc, a_sinc;
  if constexpr (OPA == OP_N) {  // [i][k] image, 16-elem rows
    constexpr int RPI = 64 * CPL / BK;  // rows per instr (c128: 4, c64: 8)
    a_l0 = (int64_t)(i0 + (NAW * w) * RPI + lane / (BK / CPL)) * lda +
           (lane % (BK / CPL)) * CPL;
    a_jinc = (int64_t)RPI * lda;
    a_sinc = BK;
  } else {  // [k][i] image, 64-elem k-rows
    constexpr int KPI = 64 * CPL / 64;  // k-rows per instr (c128: 1, c64: 2)
    a_l0 = (int64_t)(NAW * w * KPI + lane / (64 / CPL)) * lda + i0 +
           (lane % (64 / CPL)) * CPL;
    a_jinc = (int64_t)KPI * lda;
    a_sinc = (int64_t)BK * lda;
  }
  int64_t b_l0, b_jinc, b_sinc;
  if constexpr (OPB == OP_N) {
    constexpr int KPI = 64 * CPL / 64;
    b_l0 = (int64_t)(NAW * w * KPI + lane / (64 / CPL)) * ldb + j0 +
           (lane % (64 / CPL)) * CPL;
    b_jinc = (int64_t)KPI * ldb;
    b_sinc = (int64_t)BK * ldb;
  } else {
    constexpr int RPI = 64 * CPL / BK;
    b_l0 = (int64_t)(j0 + (NAW * w) * RPI + lane / (BK / CPL)) * ldb +
           (lane % (BK / CPL)) * CPL;
    b_jinc = (int64_t)RPI * ldb;
    b_sinc = BK;
  }

  const T* aP = Abase0 + 2 * a_l0;
  const T* bP = Bbase0 + 2 * b_l0;
  int ks_cnt = 0;
  int64_t kt = 0;
  T* const Sflat = &S[0][0];
  constexpr int BUFE = 4096 * (8 / sizeof(T));
  constexpr int BOFF = 2048 * (8 / sizeof(T));
  const int ldsw = w * NAW * 64 * CPL * 2;  // real-T units

  auto issue = [&](int buf) {
#pragma unroll
    for (int j = 0; j < NAW; ++j)
      glds16(aP + 2 * j * a_jinc, Sflat + buf * BUFE + ldsw + j * 64 * CPL * 2);
#pragma unroll
    for (int j = 0; j < NAW; ++j)
      glds16(bP + 2 * j * b_jinc,
             Sflat + buf * BUFE + BOFF + ldsw + j * 64 * CPL * 2);
    if (++ks_cnt == spt) {
      ks_cnt = 0;
      ++kt;
      aP = Abase0 + 2 * (kt * d.a_kstride + a_l0);
      bP = Bbase0 + 2 * (kt * d.b_kstride + b_l0);
    } else {
      aP += 2 * a_sinc;
      bP += 2 * b_sinc;
    }
  };

  auto compute = [&](int buf) {
    const T* sa = Sflat + buf * BUFE;
    const T* sb = sa + BOFF;
    auto aidx = [&](int row, int k) {
      return 2 * ((OPA == OP_N) ? row * BK + k : k * 64 + row);
    };
    auto bidx = [&](int col, int k) {
      return 2 * ((OPB == OP_N) ? k * 64 + col : col * BK + k);
    };
#pragma unroll
    for (int ks = 0; ks < BK / 4; ++ks) {
      const int k = ks * 4 + lk;
      T ar[2], ai[2], br[2], bi_[2];
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        const int ix = aidx(wrow + mi * 16 + li, k);
        ar[mi] = sa[ix];
        ai[mi] = (OPA == OP_C) ? -sa[ix + 1] : sa[ix + 1];
      }
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const int ix = bidx(wcol + ni * 16 + li, k);
        br[ni] = sb[ix];
        bi_[ni] = (OPB == OP_C) ? -sb[ix + 1] : sb[ix + 1];
      }
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          accr[mi][ni] = MfmaV2<T>::mma(ar[mi], br[ni], accr[mi][ni]);
          accr[mi][ni] = MfmaV2<T>::mma(-ai[mi], bi_[ni], accr[mi][ni]);
          acci[mi][ni] = MfmaV2<T>::mma(ar[mi], bi_[ni], acci[mi][ni]);
          acci[mi][ni] = MfmaV2<T>::mma(ai[mi], br[ni], acci[mi][ni]);
        }
    }
  };

  issue(0);
  for (int s = 0; s < total; ++s) {
    __syncthreads();
    if (s + 1 < total) issue(1 - (s & 1));
    compute(s & 1);
  }

  T* Cb = C + 2 * d.c_off;
  auto store = [&](auto betanz) {
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const acc_t vr = accr[mi][ni];
        const acc_t vi = acci[mi][ni];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = i0 + wrow + mi * 16 + MfmaV2<T>::acc_row(lk, r);
          const int col = j0 + wcol + ni * 16 + li;
          const int64_t off = 2 * ((int64_t)row * ldc + col);
          T outr = alpha_re * (T)vr[r] - alpha_im * (T)vi[r];
          T outi = alpha_re * (T)vi[r] + alpha_im * (T)vr[r];
          if constexpr (decltype(betanz)::value) {
            const T cr = Cb[off], ci = Cb[off + 1];
            outr += beta_re * cr - beta_im * ci;
            outi += beta_re * ci + beta_im * cr;
          }
          Cb[off] = outr;
          Cb[off + 1] = outi;
        }
      }
  };
  if (beta_re != T(0) || beta_im != T(0))
    store(std::true_type{});
  else
    store(std::false_type{});
}

template <typename T>
int launch_v2_real(const GemmDesc* descs, int ndesc, const T* A, const T* B,
                   T* C, int M, int N, int K, int lda, int ldb, int ldc,
                   int opA, int opB, T alpha, T beta, hipStream_t stream) {
  if (M % 128 || N % 128 || K % 16 || K <= 0) return 0;
  static const int enabled = [] {
    const char* v = getenv("DLAF_GEMM_V2");
    return v ? atoi(v) : 1;
  }();
  static const int swz = [] {
    const char* v = getenv("DLAF_GEMM_V2_SWZ");
    return v ? atoi(v) : 1;
  }();
  if (!enabled) return 0;
  const int mblocks = M / 128, nblocks = N / 128;
  const dim3 grid(ndesc * mblocks * nblocks);
  const dim3 block(256);
  const int oa = (opA == OP_C) ? OP_T : opA;
  const int ob = (opB == OP_C) ? OP_T : opB;
#define CASE(OA, OB)                                                       \
  if (oa == OA && ob == OB) {                                              \
    if (swz)                                                               \
      gemm_v2_k<T, OA, OB, 1><<<grid, block, 0, stream>>>(                 \
          descs, A, B, C, M, N, K, lda, ldb, ldc, alpha, beta, mblocks,    \
          nblocks);                                                        \
    else                                                                   \
      gemm_v2_k<T, OA, OB, 0><<<grid, block, 0, stream>>>(                 \
          descs, A, B, C, M, N, K, lda, ldb, ldc, alpha, beta, mblocks,    \
          nblocks);                                                        \
    return 1;                                                              \
  }
  CASE(OP_N, OP_N) CASE(OP_N, OP_T) CASE(OP_T, OP_N) CASE(OP_T, OP_T)
#undef CASE
  return 0;
}

template <typename T>
int launch_v2_cplx(const GemmDesc* descs, int ndesc, const T* A, const T* B,
                   T* C, int M, int N, int K, int lda, int ldb, int ldc,
                   int opA, int opB, T ar, T ai, T br, T bi,
                   hipStream_t stream) {
  if (M % 64 || N % 64 || K % 16 || K <= 0) return 0;
  static const int enabled = [] {
    const char* v = getenv("DLAF_GEMM_V2");
    return v ? atoi(v) : 1;
  }();
  static const int swz = [] {
    const char* v = getenv("DLAF_GEMM_V2_SWZ");
    return v ? atoi(v) : 1;
  }();
  if (!enabled) return 0;
  const int mblocks = M / 64, nblocks = N / 64;
  const dim3 grid(ndesc * mblocks * nblocks);
  const dim3 block(256);
#define CASE(OA, OB)                                                       \
  if (opA == OA && opB == OB) {                                            \
    if (swz)                                                               \
      gemm_v2_cplx_k<T, OA, OB, 1><<<grid, block, 0, stream>>>(            \
          descs, A, B, C, M, N, K, lda, ldb, ldc, ar, ai, br, bi, mblocks, \
          nblocks);                                                        \
    else                                                                   \
      gemm_v2_cplx_k<T, OA, OB, 0><<<grid, block, 0, stream>>>(            \
          descs, A, B, C, M, N, K, lda, ldb, ldc, ar, ai, br, bi, mblocks, \
          nblocks);                                                        \
    return 1;                                                              \
  }
  CASE(OP_N, OP_N) CASE(OP_N, OP_T) CASE(OP_N, OP_C)
  CASE(OP_T, OP_N) CASE(OP_T, OP_T) CASE(OP_T, OP_C)
  CASE(OP_C, OP_N) CASE(OP_C, OP_T) CASE(OP_C, OP_C)
#undef CASE
  return 0;
}

}  // namespace

extern "C" {

int gemm_tiles_v2_f64(const GemmDesc* descs, int ndesc, const double* A,
                      const double* B, double* C, int M, int N, int K, int lda,
                      int ldb, int ldc, int opA, int opB, double alpha,
                      double beta, hipStream_t stream) {
  return launch_v2_real<double>(descs, ndesc, A, B, C, M, N, K, lda, ldb, ldc,
                                opA, opB, alpha, beta, stream);
}

int gemm_tiles_v2_f32(const GemmDesc* descs, int ndesc, const float* A,
                      const float* B, float* C, int M, int N, int K, int lda,
                      int ldb, int ldc, int opA, int opB, float alpha,
                      float beta, hipStream_t stream) {
  return launch_v2_real<float>(descs, ndesc, A, B, C, M, N, K, lda, ldb, ldc,
                               opA, opB, alpha, beta, stream);
}

int gemm_tiles_v2_c128(const GemmDesc* descs, int ndesc, const double* A,
                       const double* B, double* C, int M, int N, int K,
                       int lda, int ldb, int ldc, int opA, int opB,
                       double alpha_re, double alpha_im, double beta_re,
                       double beta_im, hipStream_t stream) {
  return launch_v2_cplx<double>(descs, ndesc, A, B, C, M, N, K, lda, ldb, ldc,
                                opA, opB, alpha_re, alpha_im, beta_re, beta_im,
                                stream);
}

int gemm_tiles_v2_c64(const GemmDesc* descs, int ndesc, const float* A,
                      const float* B, float* C, int M, int N, int K, int lda,
                      int ldb, int ldc, int opA, int opB, float alpha_re,
                      float alpha_im, float beta_re, float beta_im,
                      hipStream_t stream) {
  return launch_v2_cplx<float>(descs, ndesc, A, B, C, M, N, K, lda, ldb, ldc,
                               opA, opB, alpha_re, alpha_im, beta_re, beta_im,
                               stream);
}

}  // extern "C"
