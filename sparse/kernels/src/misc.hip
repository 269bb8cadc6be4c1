// Misc kernels: tropical SpMV, RK stage fusion, euclidean cdist.
//
// Reference parity: CSR_SPMV_ROW_SPLIT_TROPICAL_SEMIRING
// (tropical_spmv.cu:26-56), RK_CALC_DY (runge_kutta.cu:26-42),
// EUCLIDEAN_CDIST (euclidean_distance.cu:28-61).
#include "common.h"

namespace {

// (max, lexicographic) semiring over int64 multi-field vectors
template <typename index_t>
__global__ void tropical_spmv_kernel(const int64_t* __restrict__ indptr,
                                     const index_t* __restrict__ indices,
                                     const int64_t* __restrict__ x,  // (w, nf)
                                     int64_t* __restrict__ y,        // (m, nf)
                                     int64_t m, int64_t nf, int64_t col_lo) {
  int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (r >= m) return;
  int64_t e = indptr[r + 1];
  bool any = false;
  for (int64_t p = indptr[r]; p < e; ++p) {
    const int64_t* cand = &x[((int64_t)indices[p] - col_lo) * nf];
    if (!any) {
      for (int64_t f = 0; f < nf; ++f) y[r * nf + f] = cand[f];
      any = true;
      continue;
    }
    // lexicographic compare cand vs current y row
    bool greater = false;
    for (int64_t f = 0; f < nf; ++f) {
      int64_t a = cand[f], b = y[r * nf + f];
      if (a != b) { greater = a > b; break; }
    }
    if (greater) {
      for (int64_t f = 0; f < nf; ++f) y[r * nf + f] = cand[f];
    }
  }
}

// dy[i] = h * sum_j K[j, i] * a[j]
template <typename T>
__global__ void rk_calc_dy_kernel(const T* __restrict__ K,  // (s, n) row-major
                                  const T* __restrict__ a,  // (s,)
                                  T* __restrict__ dy, int64_t n, int64_t s,
                                  T h) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  T acc = ZeroOf<T>::value();
  for (int64_t j = 0; j < s; ++j) acc += K[j * n + i] * a[j];
  dy[i] = h * acc;
}

// out[i,j] = ||XA[i,:] - XB[j,:]||_2 ; 16x16 output tiles, LDS-staged panels
template <typename T>
__global__ __launch_bounds__(256) void cdist_kernel(
    const T* __restrict__ XA, const T* __restrict__ XB, T* __restrict__ out,
    int64_t mA, int64_t mB, int64_t k) {
  __shared__ T sa[16][17];
  __shared__ T sb[16][17];
  int64_t i0 = (int64_t)blockIdx.y * 16;
  int64_t j0 = (int64_t)blockIdx.x * 16;
  int ti = threadIdx.y, tj = threadIdx.x;  // 16x16
  T acc = T(0);
  for (int64_t t = 0; t < k; t += 16) {
    int64_t ia = i0 + ti, ja = j0 + ti;
    int64_t kk = t + tj;
    sa[ti][tj] = (ia < mA && kk < k) ? XA[ia * k + kk] : T(0);
    sb[ti][tj] = (ja < mB && kk < k) ? XB[ja * k + kk] : T(0);
    __syncthreads();
    int64_t kmax = min((int64_t)16, k - t);
    for (int64_t q = 0; q < kmax; ++q) {
      T d = sa[ti][q] - sb[tj][q];
      acc += d * d;
    }
    __syncthreads();
  }
  int64_t i = i0 + ti, j = j0 + tj;
  if (i < mA && j < mB) out[i * mB + j] = sqrt(acc);
}

}  // namespace

void tropical_spmv_hip(at::Tensor indptr, at::Tensor indices, at::Tensor x,
                       at::Tensor y, int64_t col_lo) {
  int64_t m = indptr.numel() - 1;
  if (m == 0) return;
  DISPATCH_INDEX(indices.scalar_type(), "tropical", [&] {
    hipLaunchKernelGGL((tropical_spmv_kernel<index_t>), dim3((m + 255) / 256),
                       dim3(256), 0, cur_stream(), indptr.data_ptr<int64_t>(),
                       indices.data_ptr<index_t>(), x.data_ptr<int64_t>(),
                       y.data_ptr<int64_t>(), m, x.size(1), col_lo);
  });
}

void rk_calc_dy_hip(at::Tensor K, at::Tensor a, double h, at::Tensor dy) {
  int64_t n = dy.numel();
  if (n == 0) return;
  DISPATCH_VALUES(K.scalar_type(), "rk_calc_dy", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL((rk_calc_dy_kernel<T>), dim3((n + 255) / 256), dim3(256),
                       0, cur_stream(), K.data_ptr<T>(), a.data_ptr<T>(),
                       dy.data_ptr<T>(), n, K.size(0), static_cast<T>(h));
  });
}

void cdist_hip(at::Tensor XA, at::Tensor XB, at::Tensor out) {
  int64_t mA = XA.size(0), mB = XB.size(0), k = XA.size(1);
  if (mA == 0 || mB == 0) return;
  AT_DISPATCH_FLOATING_TYPES(XA.scalar_type(), "cdist", [&] {
    using T = scalar_t;
    dim3 block(16, 16);
    dim3 grid((mB + 15) / 16, (mA + 15) / 16);
    hipLaunchKernelGGL((cdist_kernel<T>), grid, block, 0, cur_stream(),
                       XA.data_ptr<T>(), XB.data_ptr<T>(), out.data_ptr<T>(),
                       mA, mB, k);
  });
}
