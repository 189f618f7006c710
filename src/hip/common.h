// SPDX-License-Identifier: Apache-2.0
// Common device helpers for legate_sparse gfx950 HIP kernels.
//
// Design per /opt/skills/guides/cdna_hip_programming.md:
//   - wave = 64 lanes (hard-coded), blocks multiple of 64
//   - memory-bound kernels: grid-stride, grid capped (G11)
//   - vectorized global access where layout permits
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>
#include <stdexcept>
#include <string>
#include <type_traits>

#define WAVE_SIZE 64
#define LS_THREADS 256

// Debug bounds checking (build with LS_DEBUG=1 -> -DLS_BOUNDS_CHECK):
// aborts the kernel with a device-side trap on an out-of-range index —
// the analogue of the reference's Legion_BOUNDS_CHECKS accessor checks
// (SURVEY §5).  Pair with AMD_SERIALIZE_KERNEL=3 to attribute the trap
// to a launch site.
#ifdef LS_BOUNDS_CHECK
#define LS_ASSERT_RANGE(i, n)                                     do {                                                              if ((i64)(i) < 0 || (i64)(i) >= (i64)(n)) {                       __builtin_trap();                                             }                                                             } while (0)
#else
#define LS_ASSERT_RANGE(i, n)   do {                          } while (0)
#endif

using i64 = int64_t;

static inline void ls_check(hipError_t e, const char* what) {
  if (e != hipSuccess) {
    throw std::runtime_error(std::string("HIP error in ") + what + ": " +
                             hipGetErrorString(e));
  }
}

static inline int grid_1d(i64 work_items, int per_block, int cap = 4096) {
  i64 blocks = (work_items + per_block - 1) / per_block;
  if (blocks < 1) blocks = 1;
  if (blocks > cap) blocks = cap;
  return static_cast<int>(blocks);
}

// ---------------------------------------------------------------------------
// Minimal complex type usable in device code (no std::complex on device).
// ---------------------------------------------------------------------------
template <typename T>
struct Cplx {
  T re, im;
  Cplx() = default;  // trivially constructible (required for __shared__)
  __host__ __device__ Cplx(T r, T i) : re(r), im(i) {}
  __host__ __device__ Cplx operator+(const Cplx& o) const {
    return {re + o.re, im + o.im};
  }
  __host__ __device__ Cplx operator-(const Cplx& o) const {
    return {re - o.re, im - o.im};
  }
  __host__ __device__ Cplx operator*(const Cplx& o) const {
    return {re * o.re - im * o.im, re * o.im + im * o.re};
  }
  __host__ __device__ Cplx operator/(const Cplx& o) const {
    T d = o.re * o.re + o.im * o.im;
    return {(re * o.re + im * o.im) / d, (im * o.re - re * o.im) / d};
  }
  __host__ __device__ Cplx& operator+=(const Cplx& o) {
    re += o.re;
    im += o.im;
    return *this;
  }
  __host__ __device__ Cplx conj() const { return {re, -im}; }
};

template <typename T>
struct is_cplx { static constexpr bool value = false; };
template <typename T>
struct is_cplx<Cplx<T>> { static constexpr bool value = true; };

template <typename T>
__device__ inline T ls_conj(T v) { return v; }
template <typename T>
__device__ inline Cplx<T> ls_conj(Cplx<T> v) { return v.conj(); }

template <typename T>
__device__ inline T ls_zero() { return T(0); }
template <>
__device__ inline Cplx<float> ls_zero<Cplx<float>>() { return {0.f, 0.f}; }
template <>
__device__ inline Cplx<double> ls_zero<Cplx<double>>() { return {0., 0.}; }

// dtype codes shared with Python: 0=f32, 1=f64, 2=c64, 3=c128
#define DISPATCH_VAL_T(code, CALL)                                  \
  switch (code) {                                                   \
    case 0: { using val_t = float;         CALL; break; }           \
    case 1: { using val_t = double;        CALL; break; }           \
    case 2: { using val_t = Cplx<float>;   CALL; break; }           \
    case 3: { using val_t = Cplx<double>;  CALL; break; }           \
    default: throw std::runtime_error("bad dtype code");            \
  }

// index dtype codes shared with Python: 0=int32, 1=int64
// (int32 column indices halve the streamed index bytes of SpMV/SpGEMM —
// the dominant optimization for memory-bound fp64 sparse kernels)
#define DISPATCH_IDX_T(code, CALL)                                  \
  switch (code) {                                                   \
    case 0: { using idx_t = int32_t; CALL; break; }                 \
    case 1: { using idx_t = int64_t; CALL; break; }                 \
    default: throw std::runtime_error("bad index dtype code");      \
  }

// ---------------------------------------------------------------------------
// Wave / block reductions (64-wide wave; __shfl_down works across 64 lanes)
// ---------------------------------------------------------------------------
template <typename T>
__device__ inline T ls_shfl_down(T v, int off) {
  return __shfl_down(v, off, WAVE_SIZE);
}
template <typename T>
__device__ inline Cplx<T> ls_shfl_down(Cplx<T> v, int off) {
  return {__shfl_down(v.re, off, WAVE_SIZE),
          __shfl_down(v.im, off, WAVE_SIZE)};
}

// Sub-group (power-of-two W <= 64, lanes grouped by lane_id / W) reduction.
template <typename T, int W>
__device__ inline T group_reduce_sum(T v) {
#pragma unroll
  for (int off = W / 2; off > 0; off >>= 1) {
    v += ls_shfl_down(v, off);
  }
  return v;
}

template <typename T>
__device__ inline T wave_reduce_sum(T v) {
  return group_reduce_sum<T, WAVE_SIZE>(v);
}

__device__ inline uint32_t wave_reduce_max_u32(uint32_t v) {
#pragma unroll
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) {
    uint32_t o = __shfl_xor(v, off, WAVE_SIZE);
    v = o > v ? o : v;
  }
  return v;
}
