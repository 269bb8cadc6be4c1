// SpMM / rSpMM / SDDMM kernels.
//
// Reference parity: SPMM_CSR_DENSE (src/sparse/array/csr/spmm.cu:26-117),
// SPMM_DENSE_CSR (spmm.cu:115-180, DISTAL nnz-split), CSR_SDDMM
// (sddmm.cu:25-85).
#include "common.h"

namespace {

// C[r, j] = sum_p vals[p] * B[col[p]-col_lo, j]; 64 lanes over j, 4 rows/block
template <typename T, typename index_t>
__global__ __launch_bounds__(256) void spmm_kernel(
    const int64_t* __restrict__ indptr, const index_t* __restrict__ indices,
    const T* __restrict__ vals, const T* __restrict__ B, T* __restrict__ C,
    int64_t m, int64_t k, int64_t col_lo) {
  int64_t r = (int64_t)blockIdx.y * blockDim.y + threadIdx.y;
  int64_t j0 = (int64_t)blockIdx.x * WAVE;
  int64_t j = j0 + threadIdx.x;
  if (r >= m) return;
  T acc = ZeroOf<T>::value();
  int64_t e = indptr[r + 1];
  for (int64_t p = indptr[r]; p < e; ++p) {
    int64_t c = (int64_t)indices[p] - col_lo;
    if (j < k) acc += vals[p] * B[c * k + j];
  }
  if (j < k) C[r * k + j] = acc;
}

// Small-k variant: lanes tile (rows x columns) so k < WAVE does not idle
// 64-k lanes (k=8 would idle 87% of the wave in spmm_kernel).  NOTE:
// nt loads on vals/indices measured 20-50% SLOWER here (unlike SpMV) —
// the same val/index address is read by all kp lanes of the wave, and the
// nontemporal policy defeats the cache broadcast that makes that free.  kp =
// next_pow2(k) lanes per row, WAVE/kp rows per wave.
template <typename T, typename index_t>
__global__ __launch_bounds__(256) void spmm_smallk_kernel(
    const int64_t* __restrict__ indptr, const index_t* __restrict__ indices,
    const T* __restrict__ vals, const T* __restrict__ B, T* __restrict__ C,
    int64_t m, int64_t k, int64_t col_lo, int kp) {
  const int lane = threadIdx.x;
  const int rpw = WAVE / kp;
  const int64_t r = ((int64_t)blockIdx.x * blockDim.y + threadIdx.y) * rpw
                    + lane / kp;
  const int64_t j = lane % kp;
  if (r >= m) return;
  T acc = ZeroOf<T>::value();
  const int64_t e = indptr[r + 1];
  for (int64_t p = indptr[r]; p < e; ++p) {
    const int64_t c = (int64_t)indices[p] - col_lo;
    if (j < k) acc += vals[p] * B[c * k + j];
  }
  if (j < k) C[r * k + j] = acc;
}

// C[i, c] += A[i, r] * v for each nz (r,c,v): one wave per nz, lanes over i
template <typename T, typename index_t>
__global__ void rspmm_kernel(const int64_t* __restrict__ indptr,
                             const index_t* __restrict__ indices,
                             const T* __restrict__ vals,
                             const T* __restrict__ A,  // (kd, mloc) row-major
                             T* __restrict__ C,        // (kd, n) row-major
                             int64_t mloc, int64_t n, int64_t kd, int64_t nnz) {
  int64_t w = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int lane = threadIdx.x % WAVE;
  if (w >= nnz) return;
  int64_t r = ub_i64(indptr, mloc + 1, w) - 1;
  int64_t c = (int64_t)indices[w];
  T v = vals[w];
  for (int64_t i = lane; i < kd; i += WAVE) {
    atomic_add_any(&C[i * n + c], A[i * mloc + r] * v);
  }
}

// out[p] = vals[p] * sum_t Cm[row,t] * D[t, col-col_lo]; D is the gathered
// COLUMN BLOCK [col_lo, col_lo+n) of the global operand (reference
// MinMaxImage on proj dim 1, csr.py:1244-1312)
template <typename T, typename index_t>
__global__ void sddmm_kernel(const int64_t* __restrict__ indptr,
                             const index_t* __restrict__ indices,
                             const T* __restrict__ vals,
                             const T* __restrict__ Cm,  // (m, kd) row-major
                             const T* __restrict__ D,   // (kd, n) row-major
                             T* __restrict__ out, int64_t m, int64_t n,
                             int64_t kd, int64_t nnz, int64_t col_lo) {
  int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (p >= nnz) return;
  int64_t r = ub_i64(indptr, m + 1, p) - 1;
  int64_t c = (int64_t)indices[p] - col_lo;
  T acc = ZeroOf<T>::value();
  for (int64_t t = 0; t < kd; ++t) acc += Cm[r * kd + t] * D[t * n + c];
  out[p] = vals[p] * acc;
}

}  // namespace

void spmm_hip(at::Tensor indptr, at::Tensor indices, at::Tensor vals,
              at::Tensor B, at::Tensor C, int64_t col_lo) {
  int64_t m = indptr.numel() - 1;
  int64_t k = B.size(1);
  if (m == 0) return;
  DISPATCH_VALUES(vals.scalar_type(), "spmm", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(indices.scalar_type(), "spmm_idx", [&] {
      if (k <= WAVE / 2) {
        int kp = 1;
        while (kp < k) kp <<= 1;
        const int rpw = WAVE / kp;
        dim3 block(WAVE, 4);
        dim3 grid((m + (int64_t)4 * rpw - 1) / ((int64_t)4 * rpw));
        hipLaunchKernelGGL((spmm_smallk_kernel<T, index_t>), grid, block, 0,
                           cur_stream(), indptr.data_ptr<int64_t>(),
                           indices.data_ptr<index_t>(), vals.data_ptr<T>(),
                           B.data_ptr<T>(), C.data_ptr<T>(), m, k, col_lo, kp);
        return;
      }
      dim3 block(WAVE, 4);
      dim3 grid((k + WAVE - 1) / WAVE, (m + 3) / 4);
      hipLaunchKernelGGL((spmm_kernel<T, index_t>), grid, block, 0, cur_stream(),
                         indptr.data_ptr<int64_t>(), indices.data_ptr<index_t>(),
                         vals.data_ptr<T>(), B.data_ptr<T>(), C.data_ptr<T>(),
                         m, k, col_lo);
    });
  });
}

void rspmm_hip(at::Tensor indptr, at::Tensor indices, at::Tensor vals,
               at::Tensor A, at::Tensor C) {
  int64_t nnz = vals.numel();
  if (nnz == 0) return;
  int64_t mloc = indptr.numel() - 1;
  int64_t kd = A.size(0);
  int64_t n = C.size(1);
  DISPATCH_VALUES(vals.scalar_type(), "rspmm", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(indices.scalar_type(), "rspmm_idx", [&] {
      int64_t threads = nnz * WAVE;
      hipLaunchKernelGGL((rspmm_kernel<T, index_t>),
                         dim3((threads + 255) / 256), dim3(256), 0, cur_stream(),
                         indptr.data_ptr<int64_t>(), indices.data_ptr<index_t>(),
                         vals.data_ptr<T>(), A.data_ptr<T>(), C.data_ptr<T>(),
                         mloc, n, kd, nnz);
    });
  });
}

void sddmm_hip(at::Tensor indptr, at::Tensor indices, at::Tensor vals,
               at::Tensor Cm, at::Tensor D, at::Tensor out, int64_t col_lo) {
  int64_t nnz = vals.numel();
  if (nnz == 0) return;
  int64_t m = indptr.numel() - 1;
  DISPATCH_VALUES(vals.scalar_type(), "sddmm", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(indices.scalar_type(), "sddmm_idx", [&] {
      hipLaunchKernelGGL((sddmm_kernel<T, index_t>), dim3((nnz + 255) / 256),
                         dim3(256), 0, cur_stream(), indptr.data_ptr<int64_t>(),
                         indices.data_ptr<index_t>(), vals.data_ptr<T>(),
                         Cm.data_ptr<T>(), D.data_ptr<T>(), out.data_ptr<T>(),
                         m, D.size(1), Cm.size(1), nnz, col_lo);
    });
  });
}
