// Elementwise CSR kernels + the fused CG axpby.
//
// Reference parity:
//  - add: ADD_CSR_CSR_NNZ / ADD_CSR_CSR (src/sparse/array/csr/add.cu) —
//    two-pointer row merge, two-phase.
//  - mult: ELEM_MULT_CSR_CSR (mult.cu:27-109) — row intersection.
//  - mult_dense: ELEM_MULT_CSR_DENSE (mult_dense.cu) — structure preserving.
//  - axpby: the fused y=(a/b)x+y / y=x+(a/b)y CG update with scalars read
//    from device memory (axpby.cu:25-42, linalg.py:479-496).
#include "common.h"

namespace {

template <typename index_t>
__global__ void add_nnz_kernel(const int64_t* __restrict__ aip,
                               const index_t* __restrict__ aix,
                               const int64_t* __restrict__ bip,
                               const index_t* __restrict__ bix,
                               int64_t* __restrict__ out, int64_t m) {
  int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (r >= m) return;
  int64_t pa = aip[r], ea = aip[r + 1];
  int64_t pb = bip[r], eb = bip[r + 1];
  int64_t c = 0;
  while (pa < ea && pb < eb) {
    index_t ca = aix[pa], cb = bix[pb];
    if (ca == cb) { ++pa; ++pb; }
    else if (ca < cb) ++pa;
    else ++pb;
    ++c;
  }
  out[r] = c + (ea - pa) + (eb - pb);
}

template <typename T, typename index_t>
__global__ void add_compute_kernel(
    const int64_t* __restrict__ aip, const index_t* __restrict__ aix,
    const T* __restrict__ av, const int64_t* __restrict__ bip,
    const index_t* __restrict__ bix, const T* __restrict__ bv,
    const int64_t* __restrict__ cip, index_t* __restrict__ cix,
    T* __restrict__ cv, int64_t m, T alpha, T beta) {
  int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (r >= m) return;
  int64_t pa = aip[r], ea = aip[r + 1];
  int64_t pb = bip[r], eb = bip[r + 1];
  int64_t o = cip[r];
  while (pa < ea && pb < eb) {
    index_t ca = aix[pa], cb = bix[pb];
    if (ca == cb) {
      cix[o] = ca; cv[o] = alpha * av[pa] + beta * bv[pb]; ++pa; ++pb;
    } else if (ca < cb) {
      cix[o] = ca; cv[o] = alpha * av[pa]; ++pa;
    } else {
      cix[o] = cb; cv[o] = beta * bv[pb]; ++pb;
    }
    ++o;
  }
  for (; pa < ea; ++pa, ++o) { cix[o] = aix[pa]; cv[o] = alpha * av[pa]; }
  for (; pb < eb; ++pb, ++o) { cix[o] = bix[pb]; cv[o] = beta * bv[pb]; }
}

template <typename index_t>
__global__ void mult_nnz_kernel(const int64_t* __restrict__ aip,
                                const index_t* __restrict__ aix,
                                const int64_t* __restrict__ bip,
                                const index_t* __restrict__ bix,
                                int64_t* __restrict__ out, int64_t m) {
  int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (r >= m) return;
  int64_t pa = aip[r], ea = aip[r + 1];
  int64_t pb = bip[r], eb = bip[r + 1];
  int64_t c = 0;
  while (pa < ea && pb < eb) {
    index_t ca = aix[pa], cb = bix[pb];
    if (ca == cb) { ++c; ++pa; ++pb; }
    else if (ca < cb) ++pa;
    else ++pb;
  }
  out[r] = c;
}

template <typename T, typename index_t>
__global__ void mult_compute_kernel(
    const int64_t* __restrict__ aip, const index_t* __restrict__ aix,
    const T* __restrict__ av, const int64_t* __restrict__ bip,
    const index_t* __restrict__ bix, const T* __restrict__ bv,
    const int64_t* __restrict__ cip, index_t* __restrict__ cix,
    T* __restrict__ cv, int64_t m) {
  int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (r >= m) return;
  int64_t pa = aip[r], ea = aip[r + 1];
  int64_t pb = bip[r], eb = bip[r + 1];
  int64_t o = cip[r];
  while (pa < ea && pb < eb) {
    index_t ca = aix[pa], cb = bix[pb];
    if (ca == cb) { cix[o] = ca; cv[o] = av[pa] * bv[pb]; ++o; ++pa; ++pb; }
    else if (ca < cb) ++pa;
    else ++pb;
  }
}

template <typename T, typename index_t>
__global__ void mult_dense_kernel(const int64_t* __restrict__ indptr,
                                  const index_t* __restrict__ indices,
                                  const T* __restrict__ vals,
                                  const T* __restrict__ D,  // (m, n) row-major
                                  T* __restrict__ out, int64_t m, int64_t n,
                                  int64_t nnz) {
  int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (p >= nnz) return;
  int64_t r = ub_i64(indptr, m + 1, p) - 1;
  out[p] = vals[p] * D[r * n + (int64_t)indices[p]];
}

// fused y = y ± (a/b) x  (isalpha) | y = x ± (a/b) y ; a,b 0-dim device.
// 16-byte vector loads/stores per thread (EPT = 16/sizeof(T) elements:
// 4 for fp32, 2 for fp64; 8B accesses run at ~0.6x the 16B rate on
// gfx950).  Flat (no grid-stride loop): measured 5865 vs 5070 GB/s for
// the strided form at 268M fp64 (tools/axpby_bench.hip).
template <typename T>
struct EptOf {
  static constexpr int value =
      (16 / sizeof(T)) >= 1 ? (int)(16 / sizeof(T)) : 1;
};

template <typename T>
struct alignas(EptOf<T>::value * sizeof(T) <= 16
                   ? EptOf<T>::value * sizeof(T)
                   : 16) VVec {
  T v[EptOf<T>::value];
};

template <typename T, bool ISALPHA, bool NEG>
__global__ __launch_bounds__(256) void axpby_kernel(
    T* __restrict__ y, const T* __restrict__ x, const T* __restrict__ a,
    const T* __restrict__ b, int64_t n) {
  constexpr int EPT = EptOf<T>::value;
  T s = (*a) / (*b);
  if (NEG) s = -s;
  const int64_t nv = n / EPT;
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < nv) {
    auto* y2 = reinterpret_cast<VVec<T>*>(y);
    auto* x2 = reinterpret_cast<const VVec<T>*>(x);
    VVec<T> yv = y2[i];
    const VVec<T> xv = x2[i];
#pragma unroll
    for (int j = 0; j < EPT; ++j) {
      yv.v[j] = ISALPHA ? (yv.v[j] + s * xv.v[j]) : (xv.v[j] + s * yv.v[j]);
    }
    y2[i] = yv;
  }
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    for (int64_t t = nv * EPT; t < n; ++t) {
      y[t] = ISALPHA ? (y[t] + s * x[t]) : (x[t] + s * y[t]);
    }
  }
}

}  // namespace

void add_nnz_hip(at::Tensor aip, at::Tensor aix, at::Tensor bip, at::Tensor bix,
                 at::Tensor out) {
  int64_t m = out.numel();
  if (m == 0) return;
  DISPATCH_INDEX(aix.scalar_type(), "add_nnz", [&] {
    hipLaunchKernelGGL((add_nnz_kernel<index_t>), dim3((m + 255) / 256), dim3(256),
                       0, cur_stream(), aip.data_ptr<int64_t>(),
                       aix.data_ptr<index_t>(), bip.data_ptr<int64_t>(),
                       bix.data_ptr<index_t>(), out.data_ptr<int64_t>(), m);
  });
}

void add_compute_hip(at::Tensor aip, at::Tensor aix, at::Tensor av,
                     at::Tensor bip, at::Tensor bix, at::Tensor bv,
                     at::Tensor cip, at::Tensor cix, at::Tensor cv,
                     double alpha, double beta) {
  int64_t m = aip.numel() - 1;
  if (m == 0) return;
  DISPATCH_VALUES(cv.scalar_type(), "add_compute", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(aix.scalar_type(), "add_compute_idx", [&] {
      hipLaunchKernelGGL((add_compute_kernel<T, index_t>), dim3((m + 255) / 256),
                         dim3(256), 0, cur_stream(), aip.data_ptr<int64_t>(),
                         aix.data_ptr<index_t>(), av.data_ptr<T>(),
                         bip.data_ptr<int64_t>(), bix.data_ptr<index_t>(),
                         bv.data_ptr<T>(), cip.data_ptr<int64_t>(),
                         cix.data_ptr<index_t>(), cv.data_ptr<T>(), m,
                         static_cast<T>(alpha), static_cast<T>(beta));
    });
  });
}

void mult_nnz_hip(at::Tensor aip, at::Tensor aix, at::Tensor bip, at::Tensor bix,
                  at::Tensor out) {
  int64_t m = out.numel();
  if (m == 0) return;
  DISPATCH_INDEX(aix.scalar_type(), "mult_nnz", [&] {
    hipLaunchKernelGGL((mult_nnz_kernel<index_t>), dim3((m + 255) / 256), dim3(256),
                       0, cur_stream(), aip.data_ptr<int64_t>(),
                       aix.data_ptr<index_t>(), bip.data_ptr<int64_t>(),
                       bix.data_ptr<index_t>(), out.data_ptr<int64_t>(), m);
  });
}

void mult_compute_hip(at::Tensor aip, at::Tensor aix, at::Tensor av,
                      at::Tensor bip, at::Tensor bix, at::Tensor bv,
                      at::Tensor cip, at::Tensor cix, at::Tensor cv) {
  int64_t m = aip.numel() - 1;
  if (m == 0) return;
  DISPATCH_VALUES(cv.scalar_type(), "mult_compute", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(aix.scalar_type(), "mult_compute_idx", [&] {
      hipLaunchKernelGGL((mult_compute_kernel<T, index_t>), dim3((m + 255) / 256),
                         dim3(256), 0, cur_stream(), aip.data_ptr<int64_t>(),
                         aix.data_ptr<index_t>(), av.data_ptr<T>(),
                         bip.data_ptr<int64_t>(), bix.data_ptr<index_t>(),
                         bv.data_ptr<T>(), cip.data_ptr<int64_t>(),
                         cix.data_ptr<index_t>(), cv.data_ptr<T>(), m);
    });
  });
}

void mult_dense_hip(at::Tensor indptr, at::Tensor indices, at::Tensor vals,
                    at::Tensor D, at::Tensor out) {
  int64_t nnz = vals.numel();
  if (nnz == 0) return;
  int64_t m = indptr.numel() - 1;
  DISPATCH_VALUES(vals.scalar_type(), "mult_dense", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(indices.scalar_type(), "mult_dense_idx", [&] {
      hipLaunchKernelGGL((mult_dense_kernel<T, index_t>), dim3((nnz + 255) / 256),
                         dim3(256), 0, cur_stream(), indptr.data_ptr<int64_t>(),
                         indices.data_ptr<index_t>(), vals.data_ptr<T>(),
                         D.data_ptr<T>(), out.data_ptr<T>(), m, D.size(1), nnz);
    });
  });
}

void axpby_hip(at::Tensor y, at::Tensor x, at::Tensor a, at::Tensor b,
               bool isalpha, bool negate) {
  int64_t n = y.numel();
  if (n == 0) return;
  const int64_t ept = std::max<int64_t>(1, 16 / y.element_size());
  int64_t blocks = (n / ept + 255) / 256 + 1;
  DISPATCH_VALUES(y.scalar_type(), "axpby", [&] {
    using T = scalar_t;
    auto launch = [&](auto kern) {
      hipLaunchKernelGGL(kern, dim3(blocks), dim3(256), 0, cur_stream(),
                         y.data_ptr<T>(), x.data_ptr<T>(), a.data_ptr<T>(),
                         b.data_ptr<T>(), n);
    };
    if (isalpha && !negate) launch(axpby_kernel<T, true, false>);
    else if (isalpha && negate) launch(axpby_kernel<T, true, true>);
    else if (!isalpha && !negate) launch(axpby_kernel<T, false, false>);
    else launch(axpby_kernel<T, false, true>);
  });
}

namespace {

// fused y-update + sum(y_new^2): the CG r-update + rz reduction in one pass
template <typename T, bool ISALPHA, bool NEG>
__global__ __launch_bounds__(256) void axpby_norm2_kernel(
    T* __restrict__ y, const T* __restrict__ x, const T* __restrict__ a,
    const T* __restrict__ b, T* __restrict__ dot_partial, int64_t n) {
  // flat 16B-per-thread + per-block partial (wrapper sums; no atomics)
  constexpr int EPT = EptOf<T>::value;
  __shared__ T red[256];
  T s = (*a) / (*b);
  if (NEG) s = -s;
  T acc = T(0);
  const int64_t nv = n / EPT;
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < nv) {
    auto* y2 = reinterpret_cast<VVec<T>*>(y);
    auto* x2 = reinterpret_cast<const VVec<T>*>(x);
    VVec<T> yv = y2[i];
    const VVec<T> xv = x2[i];
#pragma unroll
    for (int j = 0; j < EPT; ++j) {
      yv.v[j] = ISALPHA ? (yv.v[j] + s * xv.v[j]) : (xv.v[j] + s * yv.v[j]);
      acc += yv.v[j] * yv.v[j];
    }
    y2[i] = yv;
  }
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    for (int64_t t = nv * EPT; t < n; ++t) {
      T v = ISALPHA ? (y[t] + s * x[t]) : (x[t] + s * y[t]);
      y[t] = v;
      acc += v * v;
    }
  }
  red[threadIdx.x] = acc;
  __syncthreads();
  for (int w = 128; w > 0; w >>= 1) {
    if (threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
    __syncthreads();
  }
  if (threadIdx.x == 0) dot_partial[blockIdx.x] = red[0];
}

// CG K2: x += (a/b) p ; r -= (a/b) q ; per-block partials of sum(r_new^2).
// One pass over 4 streams (reads x,p,r,q; writes x,r) instead of the
// separate x-axpby + r-axpby_norm2 pair — saves 2 full HBM passes per CG
// iteration at n=268M.
template <typename T>
__global__ __launch_bounds__(256) void cg_xr_norm2_kernel(
    T* __restrict__ x, const T* __restrict__ p, T* __restrict__ r,
    const T* __restrict__ q, const T* __restrict__ a, const T* __restrict__ b,
    T* __restrict__ dot_partial, int64_t n) {
  constexpr int EPT = EptOf<T>::value;
  __shared__ T red[256];
  const T s = (*a) / (*b);
  T acc = T(0);
  const int64_t nv = n / EPT;
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < nv) {
    auto* x2 = reinterpret_cast<VVec<T>*>(x);
    auto* r2 = reinterpret_cast<VVec<T>*>(r);
    auto* p2 = reinterpret_cast<const VVec<T>*>(p);
    auto* q2 = reinterpret_cast<const VVec<T>*>(q);
    VVec<T> xv = x2[i];
    VVec<T> rv = r2[i];
    const VVec<T> pv = p2[i];
    const VVec<T> qv = q2[i];
#pragma unroll
    for (int j = 0; j < EPT; ++j) {
      xv.v[j] += s * pv.v[j];
      rv.v[j] -= s * qv.v[j];
      acc += rv.v[j] * rv.v[j];
    }
    x2[i] = xv;
    r2[i] = rv;
  }
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    for (int64_t t = nv * EPT; t < n; ++t) {
      x[t] += s * p[t];
      const T rv = r[t] - s * q[t];
      r[t] = rv;
      acc += rv * rv;
    }
  }
  red[threadIdx.x] = acc;
  __syncthreads();
  for (int w = 128; w > 0; w >>= 1) {
    if ((int)threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
    __syncthreads();
  }
  if (threadIdx.x == 0) dot_partial[blockIdx.x] = red[0];
}

}  // namespace

void cg_xr_norm2_hip(at::Tensor x, at::Tensor p, at::Tensor r, at::Tensor q,
                     at::Tensor a, at::Tensor b, at::Tensor dot_out) {
  int64_t n = x.numel();
  if (n == 0) return;
  const int64_t ept = std::max<int64_t>(1, 16 / x.element_size());
  int64_t blocks = (n / ept + 255) / 256 + 1;
  TORCH_CHECK(dot_out.numel() >= blocks,
              "cg_xr_norm2: dot_out too small (", dot_out.numel(), " < ",
              blocks, ")");
  AT_DISPATCH_FLOATING_TYPES(x.scalar_type(), "cg_xr_norm2", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL(cg_xr_norm2_kernel<T>, dim3(blocks), dim3(256), 0,
                       cur_stream(), x.data_ptr<T>(), p.data_ptr<T>(),
                       r.data_ptr<T>(), q.data_ptr<T>(), a.data_ptr<T>(),
                       b.data_ptr<T>(), dot_out.data_ptr<T>(), n);
  });
}

void axpby_norm2_hip(at::Tensor y, at::Tensor x, at::Tensor a, at::Tensor b,
                     bool isalpha, bool negate, at::Tensor dot_out) {
  int64_t n = y.numel();
  if (n == 0) return;
  // flat 16B-per-thread, same grid as axpby_hip: ceil((n/EPT)/256) + 1
  // (the +1 block only contributes a zero partial).  NO cap: a capped grid
  // left elements beyond its reach untouched (silent corruption at large
  // n — regression-tested).
  const int64_t ept = std::max<int64_t>(1, 16 / y.element_size());
  int64_t blocks = (n / ept + 255) / 256 + 1;
  TORCH_CHECK(dot_out.numel() >= blocks,
              "axpby_norm2: dot_out too small (", dot_out.numel(), " < ",
              blocks, ")");
  AT_DISPATCH_FLOATING_TYPES(y.scalar_type(), "axpby_norm2", [&] {
    using T = scalar_t;
    auto launch = [&](auto kern) {
      hipLaunchKernelGGL(kern, dim3(blocks), dim3(256), 0, cur_stream(),
                         y.data_ptr<T>(), x.data_ptr<T>(), a.data_ptr<T>(),
                         b.data_ptr<T>(), dot_out.data_ptr<T>(), n);
    };
    if (isalpha && !negate) launch(axpby_norm2_kernel<T, true, false>);
    else if (isalpha && negate) launch(axpby_norm2_kernel<T, true, true>);
    else if (!isalpha && !negate) launch(axpby_norm2_kernel<T, false, false>);
    else launch(axpby_norm2_kernel<T, false, true>);
  });
}
