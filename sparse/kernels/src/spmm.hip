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

// -- BSR MFMA SpMM ------------------------------------------------------
// 16x16-block SpMM on matrix cores: one wave computes C[brow*16.., jt*16..)
// with v_mfma_{f64,f32}_16x16x4.  Measured (profiles/MFMA_r02.md): 1.6x
// the lane-tiled kernel on the 5-diag band at k=32, 9-17x on dense-block
// structure.  Lane maps measured with tools/mfma_spmm_bench.hip --probe:
//   A[i][kk]: i=lane%16, kk=lane/16;  B[kk][j]: j=lane%16, kk=lane/16;
//   D[i][j]:  j=lane%16, i = lane/16 + 4*reg (f64) / 4*(lane/16) + reg (f32)
//   — the two families pack D rows DIFFERENTLY (both probed on gfx950).
typedef double mfma_d4 __attribute__((ext_vector_type(4)));
typedef float mfma_f4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ mfma_d4 mfma16x16x4(double a, double b, mfma_d4 c) {
  return __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, c, 0, 0, 0);
}
__device__ __forceinline__ mfma_f4 mfma16x16x4(float a, float b, mfma_f4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, c, 0, 0, 0);
}

namespace {

template <typename T>
__global__ __launch_bounds__(256) void bsr_mfma_spmm_kernel(
    const int64_t* __restrict__ bptr, const int* __restrict__ bcol,
    const T* __restrict__ bvals, const T* __restrict__ B, T* __restrict__ C,
    int64_t nbrows, int64_t mrows, int64_t k, int64_t col_lo, int64_t nwin) {
  using Acc = std::conditional_t<std::is_same_v<T, double>, mfma_d4, mfma_f4>;
  const int lane = threadIdx.x & 63;
  const int64_t tiles_j = (k + 15) >> 4;
  const int64_t wave = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  if (wave >= nbrows * tiles_j) return;
  const int64_t brow = wave / tiles_j;
  const int64_t jt = wave % tiles_j;
  const int li = lane & 15;
  const int lk = lane >> 4;
  const int64_t j = jt * 16 + li;
  Acc acc = {0, 0, 0, 0};
  const int64_t e = bptr[brow + 1];
  for (int64_t blk = bptr[brow]; blk < e; ++blk) {
    const T* Ab = bvals + blk * 256;
    const int64_t c0 = (int64_t)bcol[blk] * 16 - col_lo;
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      T a = Ab[li * 16 + kk * 4 + lk];
      const int64_t row = c0 + kk * 4 + lk;  // nwin guard: last global
      T b = (j < k && row < nwin) ? B[row * k + j] : T(0);  // block may pad
      acc = mfma16x16x4(a, b, acc);          // past n (A pad entries are 0)
    }
  }
  if (j >= k) return;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int64_t i = brow * 16 + (std::is_same_v<T, double> ? lk + 4 * r
                                                       : 4 * lk + r);
    if (i < mrows) C[i * k + j] = acc[r];
  }
}

}  // namespace (bsr)

void bsr_spmm_hip(at::Tensor bptr, at::Tensor bcol, at::Tensor bvals,
                  at::Tensor B, at::Tensor C, int64_t col_lo) {
  int64_t nbrows = bptr.numel() - 1;
  if (nbrows == 0) return;
  int64_t k = B.size(1);
  int64_t mrows = C.size(0);
  TORCH_CHECK(bvals.scalar_type() == at::kDouble ||
                  bvals.scalar_type() == at::kFloat,
              "bsr_spmm: fp32/fp64 only");
  int64_t waves = nbrows * ((k + 15) / 16);
  dim3 grid((waves * 64 + 255) / 256), block(256);
  int64_t nwin = B.size(0);
  if (bvals.scalar_type() == at::kDouble) {
    hipLaunchKernelGGL((bsr_mfma_spmm_kernel<double>), grid, block, 0,
                       cur_stream(), bptr.data_ptr<int64_t>(),
                       bcol.data_ptr<int>(), bvals.data_ptr<double>(),
                       B.data_ptr<double>(), C.data_ptr<double>(), nbrows,
                       mrows, k, col_lo, nwin);
  } else {
    hipLaunchKernelGGL((bsr_mfma_spmm_kernel<float>), grid, block, 0,
                       cur_stream(), bptr.data_ptr<int64_t>(),
                       bcol.data_ptr<int>(), bvals.data_ptr<float>(),
                       B.data_ptr<float>(), C.data_ptr<float>(), nbrows,
                       mrows, k, col_lo, nwin);
  }
}

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
