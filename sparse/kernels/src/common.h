// Shared device helpers for the sparse HIP (gfx950 / CDNA4) kernels.
//
// Design notes (MI355X):
//  - wavefront = 64 lanes; block = 256 threads (4 waves) unless stated.
//  - All hot kernels are HBM-bandwidth-bound: layouts chosen so value/index
//    streams are read fully-coalesced once; x gathers ride L2/L3.
//  - Complex arithmetic uses c10::complex<T> (device-ready in ROCm torch).
#pragma once

#include <type_traits>

#include <hip/hip_runtime.h>

#include <ATen/ATen.h>
#include <ATen/Dispatch.h>
#include <ATen/hip/HIPContext.h>
#include <c10/util/complex.h>

#define SPARSE_CHECK_HIP(cmd)                                             \
  do {                                                                    \
    hipError_t e = (cmd);                                                 \
    TORCH_CHECK(e == hipSuccess, "HIP error: ", hipGetErrorString(e));    \
  } while (0)

constexpr int WAVE = 64;

// upper_bound on a sorted int64 array: first index with arr[i] > key.
__device__ __forceinline__ int64_t ub_i64(const int64_t* arr, int64_t n, int64_t key) {
  int64_t lo = 0, hi = n;
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    if (arr[mid] <= key) lo = mid + 1; else hi = mid;
  }
  return lo;
}

// first index with arr[i] >= key.
__device__ __forceinline__ int64_t lb_i64(const int64_t* arr, int64_t n, int64_t key) {
  int64_t lo = 0, hi = n;
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    if (arr[mid] < key) lo = mid + 1; else hi = mid;
  }
  return lo;
}

// atomicAdd that also covers complex (adds parts separately — valid for sums)
template <typename T>
__device__ __forceinline__ void atomic_add_any(T* addr, T v) {
  atomicAdd(addr, v);
}
template <typename T>
__device__ __forceinline__ void atomic_add_any(c10::complex<T>* addr, c10::complex<T> v) {
  T* p = reinterpret_cast<T*>(addr);
  atomicAdd(p, v.real());
  atomicAdd(p + 1, v.imag());
}

template <typename T>
struct ZeroOf { static __device__ __forceinline__ T value() { return T(0); } };
template <typename T>
struct ZeroOf<c10::complex<T>> {
  static __device__ __forceinline__ c10::complex<T> value() { return {T(0), T(0)}; }
};

// dispatch over the two index dtypes
#define DISPATCH_INDEX(ITYPE, NAME, ...)                       \
  [&] {                                                        \
    if ((ITYPE) == at::kInt) {                                 \
      using index_t = int32_t;                                 \
      return __VA_ARGS__();                                    \
    } else {                                                   \
      using index_t = int64_t;                                 \
      return __VA_ARGS__();                                    \
    }                                                          \
  }()

#define DISPATCH_VALUES(VTYPE, NAME, ...) \
  AT_DISPATCH_FLOATING_AND_COMPLEX_TYPES(VTYPE, NAME, __VA_ARGS__)

// Bijective XCD-aware block remap (guide T1): the dispatcher places block b
// on XCD b%8; remapping gives each XCD a CONTIGUOUS chunk of the problem so
// neighbouring tiles share that XCD's private L2 (x-gather locality).
__device__ __forceinline__ int64_t xcd_swizzle(int64_t bid, int64_t nwg) {
  constexpr int64_t NXCD = 8;
  int64_t q = nwg / NXCD, rr = nwg % NXCD;
  int64_t xcd = bid % NXCD, idx = bid / NXCD;
  return (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
}

inline hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

// Non-temporal load for single-use streams (value/index planes): keeps them
// from evicting reusable lines (x windows, B rows) out of L2.  Complex
// types fall back to plain loads (builtin needs scalar/vector types).
template <typename T>
__device__ __forceinline__ T nt_load(const T* __restrict__ p) {
  if constexpr (std::is_arithmetic_v<T>) {
    return __builtin_nontemporal_load(p);
  } else {
    return *p;
  }
}
