// CSR SpMV, nnz-split with LDS-staged products — MI355X (gfx950) native.
//
// Replaces reference CSR_SPMV_ROW_SPLIT (src/sparse/array/csr/spmv.cu:25-123,
// cuSPARSE-backed there).  Design for CDNA4:
//  - each 256-thread block owns NNZ_PER_BLOCK consecutive nonzeros, so the
//    values and indices streams are read fully coalesced exactly once
//    (the kernel is HBM-bound: ~12 B/nnz for fp64+int32);
//  - products go through LDS (16 KB fp64 per block), then each thread sums
//    its rows' segments — no per-nnz atomics;
//  - rows cut by a block boundary produce one carry per block, combined by a
//    tiny fixup kernel (atomic-free main path);
//  - grid = nnz/2048 blocks >> 256 CUs, so the chip fills at any row count.
#include "common.h"

namespace {

constexpr int BLK = 256;
constexpr int VT = 8;
constexpr int64_t NNZ_PER_BLOCK = (int64_t)BLK * VT;  // 2048

template <typename T, typename index_t, bool BETA_ZERO>
__global__ __launch_bounds__(BLK) void spmv_kernel(
    const int64_t* __restrict__ indptr,  // m+1
    const index_t* __restrict__ indices,
    const T* __restrict__ vals,
    const T* __restrict__ x,  // window, indexed by (indices[p] - col_lo)
    T* __restrict__ y,
    int64_t m, int64_t nnz, int64_t col_lo, T beta,
    T* __restrict__ carry_val, int64_t* __restrict__ carry_row) {
  extern __shared__ char smem_raw[];
  T* prod = reinterpret_cast<T*>(smem_raw);
  __shared__ __align__(16) char red_raw[BLK * sizeof(T)];
  T* red = reinterpret_cast<T*>(red_raw);
  __shared__ int64_t sh_ro0, sh_ro1;

  const int64_t b = blockIdx.x;
  const int64_t s = b * NNZ_PER_BLOCK;
  const int64_t e = min(s + NNZ_PER_BLOCK, nnz);
  const int tid = threadIdx.x;

  // stage products (coalesced value/index reads)
  for (int64_t i = s + tid; i < e; i += BLK) {
    prod[i - s] = vals[i] * x[(int64_t)indices[i] - col_lo];
  }

  if (tid == 0) {
    // owned rows: first r with indptr[r] >= s .. first r with indptr[r] >= e
    // (indptr[0]==0 is row 0's start; search over indptr[0..m))
    sh_ro0 = lb_i64(indptr, m, s);
    sh_ro1 = (e == nnz) ? m : lb_i64(indptr, m, e);
  }
  __syncthreads();
  const int64_t ro0 = sh_ro0, ro1 = sh_ro1;

  // per-thread row sums (segments start at >= cend, disjoint from carry)
  for (int64_t r = ro0 + tid; r < ro1; r += BLK) {
    int64_t rs = indptr[r];
    int64_t re = min(indptr[r + 1], e);
    T acc = ZeroOf<T>::value();
    for (int64_t p = rs; p < re; ++p) acc += prod[p - s];
    if (BETA_ZERO) {
      y[r] = acc;
    } else {
      y[r] = acc + beta * y[r];
    }
  }

  // continuation carry: items [s, cend) belong to row ro0-1
  if (ro0 > 0) {
    int64_t cend = (ro0 < m) ? min(indptr[ro0], e) : e;
    if (cend > s) {
      T acc = ZeroOf<T>::value();
      for (int64_t p = s + tid; p < cend; p += BLK) acc += prod[p - s];
      red[tid] = acc;
      __syncthreads();
      for (int w = BLK / 2; w > 0; w >>= 1) {
        if (tid < w) red[tid] += red[tid + w];
        __syncthreads();
      }
      if (tid == 0) {
        carry_val[b] = red[0];
        carry_row[b] = ro0 - 1;
      }
      return;
    }
  }
  if (tid == 0) carry_row[b] = -1;
}

template <typename T>
__global__ void carry_fixup_kernel(const T* __restrict__ carry_val,
                                   const int64_t* __restrict__ carry_row,
                                   T* __restrict__ y, int64_t nblocks) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= nblocks) return;
  int64_t r = carry_row[i];
  if (r >= 0) atomic_add_any(&y[r], carry_val[i]);
}

template <typename T>
__global__ void scale_kernel(T* y, int64_t n, T beta) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) y[i] = beta * y[i];
}

}  // namespace

void spmv_hip(at::Tensor indptr, at::Tensor indices, at::Tensor values,
              at::Tensor x, at::Tensor y, int64_t col_lo, double beta) {
  const int64_t m = indptr.numel() - 1;
  const int64_t nnz = values.numel();
  auto stream = cur_stream();
  DISPATCH_VALUES(values.scalar_type(), "spmv", [&] {
    using T = scalar_t;
    T betav = static_cast<T>(beta);
    if (nnz == 0) {
      if (beta == 0.0) {
        SPARSE_CHECK_HIP(hipMemsetAsync(y.data_ptr(), 0, m * sizeof(T), stream));
      } else if (m > 0) {
        hipLaunchKernelGGL(scale_kernel<T>, dim3((m + 255) / 256), dim3(256), 0,
                           stream, y.data_ptr<T>(), m, betav);
      }
      return;
    }
    const int64_t nblocks = (nnz + NNZ_PER_BLOCK - 1) / NNZ_PER_BLOCK;
    auto carry_val = at::empty({nblocks}, values.options());
    auto carry_row = at::empty({nblocks}, indptr.options());
    size_t smem = NNZ_PER_BLOCK * sizeof(T);
    DISPATCH_INDEX(indices.scalar_type(), "spmv_idx", [&] {
      if (beta == 0.0) {
        hipLaunchKernelGGL((spmv_kernel<T, index_t, true>), dim3(nblocks),
                           dim3(BLK), smem, stream,
                           indptr.data_ptr<int64_t>(), indices.data_ptr<index_t>(),
                           values.data_ptr<T>(), x.data_ptr<T>(), y.data_ptr<T>(),
                           m, nnz, col_lo, betav,
                           carry_val.data_ptr<T>(), carry_row.data_ptr<int64_t>());
      } else {
        hipLaunchKernelGGL((spmv_kernel<T, index_t, false>), dim3(nblocks),
                           dim3(BLK), smem, stream,
                           indptr.data_ptr<int64_t>(), indices.data_ptr<index_t>(),
                           values.data_ptr<T>(), x.data_ptr<T>(), y.data_ptr<T>(),
                           m, nnz, col_lo, betav,
                           carry_val.data_ptr<T>(), carry_row.data_ptr<int64_t>());
      }
    });
    hipLaunchKernelGGL(carry_fixup_kernel<T>, dim3((nblocks + 255) / 256),
                       dim3(256), 0, stream, carry_val.data_ptr<T>(),
                       carry_row.data_ptr<int64_t>(), y.data_ptr<T>(), nblocks);
  });
}
