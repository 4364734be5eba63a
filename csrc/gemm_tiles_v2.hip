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
// BM=128 x BN (template 64/128), BK=16, 256 threads; 4 waves as a 2x2 grid
// of (64 x BN/2) tiles; NFRAG = BN/32 fragments of mfma_{f64,f32}_16x16x4.
// LDS (one __shared__ object — §5 trap 4a): S[DEPTH][A: 128*BK | B: BK*BN].
//   A image: OPA==N -> row-major [i][k] (XOR chunk swizzle, below), else
//   k-major [k][i]; B image: OPB==N -> k-major [k][c], else row-major
//   [c][k] (swizzled). Images are glds-lane-linear: instr t covers LDS
//   elems t*64*EPL..+64*EPL-1.
//
// Swizzle: a row-major image read column-wise is up to 8-way bank
// conflicted (rows are 128 B = 2 banks apart). The 16-B chunk at (row, c)
// holds global chunk c ^ (row & (CPR-1)) — applied on the per-lane glds
// SOURCE address (LDS stays lane-linear, the glds contract) and undone in
// the fragment index; residual conflict is 2-way.
template <typename T, int OPA, int OPB, int SWZ, int DEPTH, int BN,
          int BK = 16, int BM = 128>
__launch_bounds__(256, DEPTH == 2 ? 2 : 1) __global__ void gemm_v2_k(
    const GemmDesc* __restrict__ descs, const T* __restrict__ A,
    const T* __restrict__ B, T* __restrict__ C, int M, int N, int K, int lda,
    int ldb, int ldc, T alpha, T beta, int mblocks, int nblocks) {
  constexpr int EPL = 16 / sizeof(T);  // elements per lane per glds
  constexpr int CPR = BK / EPL;        // 16-B chunks per rk-image row
  constexpr int AEL = BM * BK;         // A image elements
  constexpr int BEL = BK * BN;         // B image elements
  constexpr int NFRAG = BN / 32;  // col frags per wave
  constexpr int MFRAG = BM / 32;  // row frags per wave
  __shared__ T S[DEPTH][AEL + BEL];

  int wg = blockIdx.x;
  if constexpr (SWZ) wg = xcd_remap(wg, gridDim.x);
  const int per_desc = mblocks * nblocks;
  const GemmDesc d = descs[wg / per_desc];
  const int rem = wg % per_desc;
  const int i0 = (rem / nblocks) * BM, j0 = (rem % nblocks) * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63, w = tid >> 6;
  const int wrow = (w >> 1) * (BM / 2), wcol = (w & 1) * (BN / 2);
  const int li = lane & 15, lk = lane >> 4;

  const int spt = K / BK;  // steps per K-tile
  const int total = (int)d.ktiles * spt;
  // glds instructions per wave per step, per operand
  constexpr int NAW = AEL / (64 * EPL) / 4;
  constexpr int NBW = BEL / (64 * EPL) / 4;

  using acc_t = typename MfmaV2<T>::acc_t;
  acc_t acc[MFRAG][NFRAG];
#pragma unroll
  for (int a = 0; a < MFRAG; ++a)
#pragma unroll
    for (int b = 0; b < NFRAG; ++b) acc[a][b] = {0, 0, 0, 0};

  // per-lane glds source offsets for instr j=0 of this wave (element units,
  // relative to the K-tile base) and the per-instr / per-step increments
  const T* Abase0 = A + d.a_off;
  const T* Bbase0 = B + d.b_off;
  int64_t a_l0, a_jinc, a_sinc;
  if constexpr (OPA == OP_N) {  // swizzled rk image [i][k]
    constexpr int RPI = 64 / CPR;  // rows per instr
    const int r_in = lane / CPR;   // row within instr; also the swizzle mask
    a_l0 = (int64_t)(i0 + (NAW * w) * RPI + r_in) * lda +
           EPL * ((lane % CPR) ^ (r_in & (CPR - 1)));
    a_jinc = (int64_t)RPI * lda;
    a_sinc = BK;
  } else {  // kr image [k][BM]
    constexpr int KPI = 64 * EPL / BM;
    a_l0 = (int64_t)(NAW * w * KPI + lane / (BM / EPL)) * lda + i0 +
           (lane % (BM / EPL)) * EPL;
    a_jinc = (int64_t)KPI * lda;
    a_sinc = (int64_t)BK * lda;
  }
  int64_t b_l0, b_jinc, b_sinc;
  if constexpr (OPB == OP_N) {  // kr image [k][BN]
    constexpr int KPI = 64 * EPL / BN;
    b_l0 = (int64_t)(NBW * w * KPI + lane / (BN / EPL)) * ldb + j0 +
           (lane % (BN / EPL)) * EPL;
    b_jinc = (int64_t)KPI * ldb;
    b_sinc = (int64_t)BK * ldb;
  } else {  // swizzled rk image [c][k]
    constexpr int RPI = 64 / CPR;
    const int r_in = lane / CPR;
    b_l0 = (int64_t)(j0 + (NBW * w) * RPI + r_in) * ldb +
           EPL * ((lane % CPR) ^ (r_in & (CPR - 1)));
    b_jinc = (int64_t)RPI * ldb;
    b_sinc = BK;
  }

  // running state of the issue pipeline
  const T* aP = Abase0 + a_l0;
  const T* bP = Bbase0 + b_l0;
  int ks_cnt = 0;
  int64_t kt = 0;
  T* const Sflat = &S[0][0];
  constexpr int BUFE = AEL + BEL;
  const int ldsa = w * NAW * 64 * EPL;  // this wave's A LDS offset (j=0)
  const int ldsb = AEL + w * NBW * 64 * EPL;

  auto issue = [&](int buf) {
#pragma unroll
    for (int j = 0; j < NAW; ++j)
      glds16(aP + j * a_jinc, Sflat + buf * BUFE + ldsa + j * 64 * EPL);
#pragma unroll
    for (int j = 0; j < NBW; ++j)
      glds16(bP + j * b_jinc, Sflat + buf * BUFE + ldsb + j * 64 * EPL);
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
    const T* sb = sa + AEL;
    // fragment index helpers (undo the rk swizzle)
    auto aidx = [&](int row, int k) {
      return (OPA == OP_N)
                 ? row * BK + EPL * ((k / EPL) ^ (row & (CPR - 1))) + k % EPL
                 : k * BM + row;
    };
    auto bidx = [&](int col, int k) {
      return (OPB == OP_N)
                 ? k * BN + col
                 : col * BK + EPL * ((k / EPL) ^ (col & (CPR - 1))) + k % EPL;
    };
    T af[2][MFRAG], bf[2][NFRAG];
#pragma unroll
    for (int mi = 0; mi < MFRAG; ++mi)
      af[0][mi] = sa[aidx(wrow + mi * 16 + li, lk)];
#pragma unroll
    for (int ni = 0; ni < NFRAG; ++ni)
      bf[0][ni] = sb[bidx(wcol + ni * 16 + li, lk)];
#pragma unroll
    for (int ks = 0; ks < BK / 4; ++ks) {
      const int cur = ks & 1;
      if (ks + 1 < BK / 4) {
        const int nxt = 1 - cur;
        const int k = (ks + 1) * 4 + lk;
#pragma unroll
        for (int mi = 0; mi < MFRAG; ++mi)
          af[nxt][mi] = sa[aidx(wrow + mi * 16 + li, k)];
#pragma unroll
        for (int ni = 0; ni < NFRAG; ++ni)
          bf[nxt][ni] = sb[bidx(wcol + ni * 16 + li, k)];
      }
#pragma unroll
      for (int mi = 0; mi < MFRAG; ++mi)
#pragma unroll
        for (int ni = 0; ni < NFRAG; ++ni)
          acc[mi][ni] = MfmaV2<T>::mma(af[cur][mi], bf[cur][ni], acc[mi][ni]);
    }
  };

  // pipeline: DEPTH buffers, DEPTH-1 steps in flight. DEPTH==2 uses plain
  // __syncthreads() (its vmcnt(0) drain IS the completion wait); DEPTH>=3
  // uses counted per-wave vmcnt waits + raw barriers so DEPTH-2 steps stay
  // in flight across each barrier (the guide's 3-buf span pattern).
  constexpr int GPW = NAW + NBW;  // glds per wave per step
  int ibuf = 0, cbuf = 0;
  for (int p = 0; p < DEPTH - 1 && p < total; ++p) {
    issue(ibuf);
    if (++ibuf == DEPTH) ibuf = 0;
  }
  for (int s = 0; s < total; ++s) {
    if constexpr (DEPTH == 2) {
      __syncthreads();  // drains the in-flight glds (vmcnt 0) + LDS reuse
    } else {
      const int rem_s = total - s - 1;  // steps still to land after this one
      if (rem_s >= DEPTH - 2)
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(GPW * (DEPTH - 2)) : "memory");
      else if (rem_s == 1)
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(GPW) : "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
    if (s + DEPTH - 1 < total) {
      issue(ibuf);
      if (++ibuf == DEPTH) ibuf = 0;
    }
    compute(cbuf);
    if (++cbuf == DEPTH) cbuf = 0;
  }

  // epilogue: C = alpha*acc + beta*C. The beta path batches its C loads per
  // mi-chunk (loads, then uses) — interleaved load/use would eat a full
  // vmcnt(0) drain per element (hipcc is conservative about ordinary loads
  // in a function that issued glds).
  T* Cb = C + d.c_off;
  if (beta != T(0)) {
#pragma unroll
    for (int mi = 0; mi < MFRAG; ++mi) {
      T cv[4 * NFRAG];
#pragma unroll
      for (int ni = 0; ni < NFRAG; ++ni)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = i0 + wrow + mi * 16 + MfmaV2<T>::acc_row(lk, r);
          const int col = j0 + wcol + ni * 16 + li;
          cv[ni * 4 + r] = Cb[(int64_t)row * ldc + col];
        }
#pragma unroll
      for (int ni = 0; ni < NFRAG; ++ni) {
        const acc_t v = acc[mi][ni];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = i0 + wrow + mi * 16 + MfmaV2<T>::acc_row(lk, r);
          const int col = j0 + wcol + ni * 16 + li;
          Cb[(int64_t)row * ldc + col] = alpha * (T)v[r] + beta * cv[ni * 4 + r];
        }
      }
    }
  } else {
#pragma unroll
    for (int mi = 0; mi < MFRAG; ++mi)
#pragma unroll
      for (int ni = 0; ni < NFRAG; ++ni) {
        const acc_t v = acc[mi][ni];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = i0 + wrow + mi * 16 + MfmaV2<T>::acc_row(lk, r);
          const int col = j0 + wcol + ni * 16 + li;
          Cb[(int64_t)row * ldc + col] = alpha * (T)v[r];
        }
      }
  }
}

// ---------------- complex kernel ----------------
// Interleaved (re,im); one lane's 16-B glds = one complex f64 element (f32:
// two). BM=BN=64, BK=16; wave w owns the 32x32 quadrant; 2x2 fragments,
// 4 MFMA per fragment pair. LDS images mirror the real kernel with complex
// elements; fragments are read as (re,im) pairs in one ds_read_b128 (f64).
template <typename T, int OPA, int OPB, int SWZ, int DEPTH = 2>
__launch_bounds__(256, DEPTH == 2 ? 2 : 1) __global__ void gemm_v2_cplx_k(
    const GemmDesc* __restrict__ descs, const T* __restrict__ A,
    const T* __restrict__ B, T* __restrict__ C, int M, int N, int K, int lda,
    int ldb, int ldc, T alpha_re, T alpha_im, T beta_re, T beta_im,
    int mblocks, int nblocks) {
  constexpr int BK = 16;
  constexpr int CPL = 8 / sizeof(T);  // complex elems per lane per glds
  // S[buf][A: 64*16 | B: 16*64] complex elems, interleaved re/im
  __shared__ T S[DEPTH][4096 * (8 / sizeof(T))];

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

  constexpr int GPW = 2 * NAW;
  int ibuf = 0, cbuf = 0;
  for (int p = 0; p < DEPTH - 1 && p < total; ++p) {
    issue(ibuf);
    if (++ibuf == DEPTH) ibuf = 0;
  }
  for (int s = 0; s < total; ++s) {
    if constexpr (DEPTH == 2) {
      __syncthreads();
    } else {
      const int rem_s = total - s - 1;
      if (rem_s >= DEPTH - 2)
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(GPW * (DEPTH - 2)) : "memory");
      else if (rem_s == 1)
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(GPW) : "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
    if (s + DEPTH - 1 < total) {
      issue(ibuf);
      if (++ibuf == DEPTH) ibuf = 0;
    }
    compute(cbuf);
    if (++cbuf == DEPTH) cbuf = 0;
  }

  T* Cb = C + 2 * d.c_off;
  if (beta_re != T(0) || beta_im != T(0)) {
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
      T cv[16];
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = i0 + wrow + mi * 16 + MfmaV2<T>::acc_row(lk, r);
          const int col = j0 + wcol + ni * 16 + li;
          const int64_t off = 2 * ((int64_t)row * ldc + col);
          cv[(ni * 4 + r) * 2] = Cb[off];
          cv[(ni * 4 + r) * 2 + 1] = Cb[off + 1];
        }
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const acc_t vr = accr[mi][ni];
        const acc_t vi = acci[mi][ni];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = i0 + wrow + mi * 16 + MfmaV2<T>::acc_row(lk, r);
          const int col = j0 + wcol + ni * 16 + li;
          const int64_t off = 2 * ((int64_t)row * ldc + col);
          const T cr = cv[(ni * 4 + r) * 2], ci = cv[(ni * 4 + r) * 2 + 1];
          Cb[off] = alpha_re * (T)vr[r] - alpha_im * (T)vi[r] +
                    beta_re * cr - beta_im * ci;
          Cb[off + 1] = alpha_re * (T)vi[r] + alpha_im * (T)vr[r] +
                        beta_re * ci + beta_im * cr;
        }
      }
    }
  } else {
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
          Cb[off] = alpha_re * (T)vr[r] - alpha_im * (T)vi[r];
          Cb[off + 1] = alpha_re * (T)vi[r] + alpha_im * (T)vr[r];
        }
      }
  }
}

template <typename T>
int launch_v2_real(const GemmDesc* descs, int ndesc, const T* A, const T* B,
                   T* C, int M, int N, int K, int lda, int ldb, int ldc,
                   int opA, int opB, T alpha, T beta, hipStream_t stream) {
  static const int bm_env = [] {
    const char* v = getenv("DLAF_GEMM_V2_BM");
    return v ? atoi(v) : 128;
  }();
  const int bm = (bm_env == 64 && M % 64 == 0) ? 64 : 128;
  if (M % bm || N % 64 || K % 16 || K <= 0) return 0;  // BK=8 needs K%16 too
  static const int enabled = [] {
    const char* v = getenv("DLAF_GEMM_V2");
    return v ? atoi(v) : 1;
  }();
  static const int swz = [] {
    const char* v = getenv("DLAF_GEMM_V2_SWZ");
    return v ? atoi(v) : 1;
  }();
  static const int depth = [] {
    const char* v = getenv("DLAF_GEMM_V2_DEPTH");
    const int d = v ? atoi(v) : 2;
    return d < 2 ? 2 : (d > 3 ? 3 : d);
  }();
  static const int bn_env = [] {
    const char* v = getenv("DLAF_GEMM_V2_BN");
    return v ? atoi(v) : 128;
  }();
  static const int bk_env = [] {
    const char* v = getenv("DLAF_GEMM_V2_BK");
    return v ? atoi(v) : 16;
  }();
  if (!enabled) return 0;
  const int bn = (N % bn_env == 0) ? bn_env : (N % 64 == 0 ? 64 : 128);
  if (N % bn) return 0;
  const int mblocks = M / bm, nblocks = N / bn;
  const dim3 grid(ndesc * mblocks * nblocks);
  const dim3 block(256);
  const int oa = (opA == OP_C) ? OP_T : opA;
  const int ob = (opB == OP_C) ? OP_T : opB;
#define LV2(OA, OB, SW, DP, BNv)                                           \
  gemm_v2_k<T, OA, OB, SW, DP, BNv><<<grid, block, 0, stream>>>(           \
      descs, A, B, C, M, N, K, lda, ldb, ldc, alpha, beta, mblocks,        \
      nblocks)
#define LV2K(OA, OB, SW, DP, BNv, BKv)                                    \
  gemm_v2_k<T, OA, OB, SW, DP, BNv, BKv><<<grid, block, 0, stream>>>(      \
      descs, A, B, C, M, N, K, lda, ldb, ldc, alpha, beta, mblocks,        \
      nblocks)
#define LV2M(OA, OB, SW)                                                   \
  gemm_v2_k<T, OA, OB, SW, 2, 128, 16, 64><<<grid, block, 0, stream>>>(    \
      descs, A, B, C, M, N, K, lda, ldb, ldc, alpha, beta, mblocks,        \
      nblocks)
#define CASE(OA, OB)                                                       \
  if (oa == OA && ob == OB) {                                              \
    if (bm == 64) {                                                        \
      if (N % 128) return 0;                                               \
      if (swz) LV2M(OA, OB, 1); else LV2M(OA, OB, 0);                      \
    } else if (bk_env == 8 && bn == 64) {                                         \
      if (swz) LV2K(OA, OB, 1, 2, 64, 8); else LV2K(OA, OB, 0, 2, 64, 8);  \
    } else if (bn == 64) {                                                 \
      if (depth == 2) {                                                    \
        if (swz) LV2(OA, OB, 1, 2, 64); else LV2(OA, OB, 0, 2, 64);        \
      } else {                                                             \
        if (swz) LV2(OA, OB, 1, 3, 64); else LV2(OA, OB, 0, 3, 64);        \
      }                                                                    \
    } else {                                                               \
      if (depth == 2) {                                                    \
        if (swz) LV2(OA, OB, 1, 2, 128); else LV2(OA, OB, 0, 2, 128);      \
      } else {                                                             \
        if (swz) LV2(OA, OB, 1, 3, 128); else LV2(OA, OB, 0, 3, 128);      \
      }                                                                    \
    }                                                                      \
    return 1;                                                              \
  }
  CASE(OP_N, OP_N) CASE(OP_N, OP_T) CASE(OP_T, OP_N) CASE(OP_T, OP_T)
#undef CASE
#undef LV2M
#undef LV2K
#undef LV2
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
  static const int depth = [] {
    const char* v = getenv("DLAF_GEMM_V2_DEPTH");
    const int d = v ? atoi(v) : 2;
    return d < 2 ? 2 : (d > 4 ? 4 : d);
  }();
  if (!enabled) return 0;
  const int mblocks = M / 64, nblocks = N / 64;
  const dim3 grid(ndesc * mblocks * nblocks);
  const dim3 block(256);
#define LV2C(OA, OB, SW, DP)                                               \
  gemm_v2_cplx_k<T, OA, OB, SW, DP><<<grid, block, 0, stream>>>(           \
      descs, A, B, C, M, N, K, lda, ldb, ldc, ar, ai, br, bi, mblocks,     \
      nblocks)
#define CASE(OA, OB)                                                       \
  if (opA == OA && opB == OB) {                                            \
    if (depth == 2) {                                                      \
      if (swz) LV2C(OA, OB, 1, 2); else LV2C(OA, OB, 0, 2);                \
    } else if (depth == 3) {                                               \
      if (swz) LV2C(OA, OB, 1, 3); else LV2C(OA, OB, 0, 3);                \
    } else {                                                               \
      if (swz) LV2C(OA, OB, 1, 4); else LV2C(OA, OB, 0, 4);                \
    }                                                                      \
    return 1;                                                              \
  }
  CASE(OP_N, OP_N) CASE(OP_N, OP_T) CASE(OP_N, OP_C)
  CASE(OP_T, OP_N) CASE(OP_T, OP_T) CASE(OP_T, OP_C)
  CASE(OP_C, OP_N) CASE(OP_C, OP_T) CASE(OP_C, OP_C)
#undef CASE
#undef LV2C
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
