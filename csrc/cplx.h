// Minimal device complex type over interleaved (re, im) storage.
#pragma once
#include <hip/hip_runtime.h>

template <typename T>
struct cplx {
  T re, im;
  cplx() = default;
  __device__ __host__ cplx(T r, T i) : re(r), im(i) {}
  __device__ inline cplx operator+(const cplx& o) const { return {re + o.re, im + o.im}; }
  __device__ inline cplx operator-(const cplx& o) const { return {re - o.re, im - o.im}; }
  __device__ inline cplx operator*(const cplx& o) const {
    return {re * o.re - im * o.im, re * o.im + im * o.re};
  }
  __device__ inline cplx operator*(T s) const { return {re * s, im * s}; }
  __device__ inline cplx operator-() const { return {-re, -im}; }
  __device__ inline cplx& operator+=(const cplx& o) {
    re += o.re;
    im += o.im;
    return *this;
  }
  __device__ inline cplx& operator-=(const cplx& o) {
    re -= o.re;
    im -= o.im;
    return *this;
  }
  __device__ inline cplx conj() const { return {re, -im}; }
  __device__ inline T abs2() const { return re * re + im * im; }
  __device__ inline cplx recip() const {
    T d = T(1) / (re * re + im * im);
    return {re * d, -im * d};
  }
};

template <typename S>
struct ScalarTraits {  // real
  using real_t = S;
  static __device__ inline S conj(S v) { return v; }
  static __device__ inline S recip(S v) { return S(1) / v; }
  static __device__ inline S real(S v) { return v; }
  static __device__ inline S from_real(real_t v) { return v; }
  static __device__ inline S zero() { return S(0); }
};
template <typename T>
struct ScalarTraits<cplx<T>> {
  using real_t = T;
  static __device__ inline cplx<T> conj(cplx<T> v) { return v.conj(); }
  static __device__ inline cplx<T> recip(cplx<T> v) { return v.recip(); }
  static __device__ inline T real(cplx<T> v) { return v.re; }
  static __device__ inline cplx<T> from_real(T v) { return {v, T(0)}; }
  static __device__ inline cplx<T> zero() { return {T(0), T(0)}; }
};
