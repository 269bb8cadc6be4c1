// Conversions + CSC col-split kernels.
//
// Reference parity: CSR_TO_DENSE (csr_to_dense.cu), CSR_DIAGONAL
// (get_diagonal.cu), CSC_SPMV_COL_SPLIT (csc/spmv.cu:60-75), SPMM_CSC_DENSE
// (csc/spmm.cu).
#include "common.h"

namespace {

template <typename T, typename index_t>
__global__ void csr_to_dense_kernel(const int64_t* __restrict__ indptr,
                                    const index_t* __restrict__ indices,
                                    const T* __restrict__ vals,
                                    T* __restrict__ out, int64_t m, int64_t n,
                                    int64_t nnz) {
  int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (p >= nnz) return;
  int64_t r = ub_i64(indptr, m + 1, p) - 1;
  out[r * n + (int64_t)indices[p]] = vals[p];
}

template <typename T, typename index_t>
__global__ void csr_diagonal_kernel(const int64_t* __restrict__ indptr,
                                    const index_t* __restrict__ indices,
                                    const T* __restrict__ vals,
                                    T* __restrict__ out, int64_t m,
                                    int64_t row_offset) {
  int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (r >= m) return;
  int64_t target = r + row_offset;
  int64_t lo = indptr[r], hi = indptr[r + 1];
  // binary search in sorted row for column == target
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    int64_t c = (int64_t)indices[mid];
    if (c < target) lo = mid + 1;
    else hi = mid;
  }
  if (lo < indptr[r + 1] && (int64_t)indices[lo] == target) out[r] = vals[lo];
}

// y[row - rlo] += v * x[col]; one thread per COLUMN (no per-nnz search —
// the scatter targets absorb the atomics; a 1B-nnz GMG restriction was
// bound by the old per-nnz binary search)
template <typename T, typename index_t>
__global__ void csc_spmv_kernel(const int64_t* __restrict__ colptr,
                                const index_t* __restrict__ rowidx,
                                const T* __restrict__ vals,
                                const T* __restrict__ x,
                                T* __restrict__ y, int64_t ncl, int64_t rlo,
                                int64_t nnz) {
  int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= ncl) return;
  const T xv = x[c];
  const int64_t e = colptr[c + 1];
  for (int64_t p = colptr[c]; p < e; ++p) {
    atomic_add_any(&y[(int64_t)rowidx[p] - rlo], vals[p] * xv);
  }
}

// C[(row-rlo), j] += v * B[col, j]; one wave per column, lanes over j
template <typename T, typename index_t>
__global__ void csc_spmm_kernel(const int64_t* __restrict__ colptr,
                                const index_t* __restrict__ rowidx,
                                const T* __restrict__ vals,
                                const T* __restrict__ B, T* __restrict__ C,
                                int64_t ncl, int64_t rlo, int64_t k,
                                int64_t nnz) {
  int64_t c = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int lane = threadIdx.x % WAVE;
  if (c >= ncl) return;
  const int64_t e = colptr[c + 1];
  for (int64_t p = colptr[c]; p < e; ++p) {
    int64_t r = (int64_t)rowidx[p] - rlo;
    T v = vals[p];
    for (int64_t j = lane; j < k; j += WAVE) {
      atomic_add_any(&C[r * k + j], v * B[c * k + j]);
    }
  }
}

}  // namespace

void csr_to_dense_hip(at::Tensor indptr, at::Tensor indices, at::Tensor vals,
                      at::Tensor out) {
  int64_t nnz = vals.numel();
  if (nnz == 0) return;
  DISPATCH_VALUES(vals.scalar_type(), "csr_to_dense", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(indices.scalar_type(), "csr_to_dense_idx", [&] {
      hipLaunchKernelGGL((csr_to_dense_kernel<T, index_t>),
                         dim3((nnz + 255) / 256), dim3(256), 0, cur_stream(),
                         indptr.data_ptr<int64_t>(), indices.data_ptr<index_t>(),
                         vals.data_ptr<T>(), out.data_ptr<T>(),
                         indptr.numel() - 1, out.size(1), nnz);
    });
  });
}

void csr_diagonal_hip(at::Tensor indptr, at::Tensor indices, at::Tensor vals,
                      at::Tensor out, int64_t row_offset) {
  int64_t m = out.numel();
  if (m == 0) return;
  DISPATCH_VALUES(vals.scalar_type(), "csr_diagonal", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(indices.scalar_type(), "csr_diagonal_idx", [&] {
      hipLaunchKernelGGL((csr_diagonal_kernel<T, index_t>),
                         dim3((m + 255) / 256), dim3(256), 0, cur_stream(),
                         indptr.data_ptr<int64_t>(), indices.data_ptr<index_t>(),
                         vals.data_ptr<T>(), out.data_ptr<T>(), m, row_offset);
    });
  });
}

void csc_spmv_hip(at::Tensor colptr, at::Tensor rowidx, at::Tensor vals,
                  at::Tensor x, at::Tensor y, int64_t rlo) {
  int64_t nnz = vals.numel();
  if (nnz == 0) return;
  DISPATCH_VALUES(vals.scalar_type(), "csc_spmv", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(rowidx.scalar_type(), "csc_spmv_idx", [&] {
      int64_t ncl = colptr.numel() - 1;
      hipLaunchKernelGGL((csc_spmv_kernel<T, index_t>), dim3((ncl + 256) / 256),
                         dim3(256), 0, cur_stream(), colptr.data_ptr<int64_t>(),
                         rowidx.data_ptr<index_t>(), vals.data_ptr<T>(),
                         x.data_ptr<T>(), y.data_ptr<T>(), colptr.numel() - 1,
                         rlo, nnz);
    });
  });
}

void csc_spmm_hip(at::Tensor colptr, at::Tensor rowidx, at::Tensor vals,
                  at::Tensor B, at::Tensor C, int64_t rlo) {
  int64_t nnz = vals.numel();
  if (nnz == 0) return;
  DISPATCH_VALUES(vals.scalar_type(), "csc_spmm", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(rowidx.scalar_type(), "csc_spmm_idx", [&] {
      int64_t threads = (colptr.numel() - 1) * WAVE;
      hipLaunchKernelGGL((csc_spmm_kernel<T, index_t>),
                         dim3((threads + 256) / 256), dim3(256), 0, cur_stream(),
                         colptr.data_ptr<int64_t>(), rowidx.data_ptr<index_t>(),
                         vals.data_ptr<T>(), B.data_ptr<T>(), C.data_ptr<T>(),
                         colptr.numel() - 1, rlo, B.size(1), nnz);
    });
  });
}

// -- segmented COO -> CSR (de-torch the conversion path) --
// Replaces the global radix sort (rocprim onesweep was ~25% of bench GPU
// time in profiles/cg_nx16384_nt_r01.md) with: per-row atomic scatter into
// precomputed indptr slots + a per-row odd-even LDS sort.  Rows longer
// than COO2CSR_MAXROW or containing duplicate columns raise a flag and
// the caller falls back to the torch sort path (rare).
namespace {

constexpr int COO2CSR_MAXROW = 1024;

template <typename T, typename index_t>
__global__ void coo_scatter_kernel(const int64_t* __restrict__ rows,
                                   const index_t* __restrict__ cols,
                                   const T* __restrict__ vals,
                                   int64_t* __restrict__ cursor,
                                   index_t* __restrict__ out_idx,
                                   T* __restrict__ out_vals, int64_t nnz) {
  int64_t e = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (e >= nnz) return;
  int64_t pos = atomicAdd(
      reinterpret_cast<unsigned long long*>(&cursor[rows[e]]), 1ull);
  out_idx[pos] = cols[e];
  out_vals[pos] = vals[e];
}

// one WAVE per block per row: LDS odd-even transposition sort of
// (indices, vals); block=64 so early exit never strands a barrier
template <typename T, typename index_t>
__global__ __launch_bounds__(64) void row_sort_kernel(
    const int64_t* __restrict__ indptr, index_t* __restrict__ indices,
    T* __restrict__ vals, int64_t m, int* __restrict__ flags) {
  __shared__ index_t sc[COO2CSR_MAXROW];
  // raw storage: c10::complex has a ctor, which __shared__ forbids
  __shared__ __align__(16) unsigned char sv_raw[COO2CSR_MAXROW * sizeof(T)];
  T* sv = reinterpret_cast<T*>(sv_raw);
  int64_t row = blockIdx.x;
  if (row >= m) return;
  const int64_t s = indptr[row];
  const int len = (int)(indptr[row + 1] - s);
  if (len <= 1) return;
  if (len > COO2CSR_MAXROW) {
    if (threadIdx.x == 0) atomicOr(&flags[0], 1);
    return;
  }
  const int lane = threadIdx.x;
  for (int i = lane; i < len; i += 64) {
    sc[i] = indices[s + i];
    sv[i] = vals[s + i];
  }
  __syncthreads();
  for (int pass = 0; pass < len; ++pass) {
    const int start = pass & 1;
    for (int t = lane; 2 * t + start + 1 < len; t += 64) {
      const int a = 2 * t + start;
      if (sc[a] > sc[a + 1]) {
        index_t tc = sc[a]; sc[a] = sc[a + 1]; sc[a + 1] = tc;
        T tv = sv[a]; sv[a] = sv[a + 1]; sv[a + 1] = tv;
      }
    }
    __syncthreads();
  }
  int dup = 0;
  for (int i = lane; i + 1 < len; i += 64) dup |= (sc[i] == sc[i + 1]);
  if (dup) atomicOr(&flags[1], 1);
  for (int i = lane; i < len; i += 64) {
    indices[s + i] = sc[i];
    vals[s + i] = sv[i];
  }
}

}  // namespace

void coo_to_csr_hip(at::Tensor rows, at::Tensor cols, at::Tensor vals,
                    at::Tensor cursor, at::Tensor indptr, at::Tensor out_idx,
                    at::Tensor out_vals, at::Tensor flags) {
  int64_t nnz = vals.numel();
  int64_t m = indptr.numel() - 1;
  if (m == 0) return;
  DISPATCH_VALUES(vals.scalar_type(), "coo_to_csr", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(cols.scalar_type(), "coo_to_csr_idx", [&] {
      if (nnz) {
        hipLaunchKernelGGL((coo_scatter_kernel<T, index_t>),
                           dim3((nnz + 255) / 256), dim3(256), 0, cur_stream(),
                           rows.data_ptr<int64_t>(), cols.data_ptr<index_t>(),
                           vals.data_ptr<T>(), cursor.data_ptr<int64_t>(),
                           out_idx.data_ptr<index_t>(), out_vals.data_ptr<T>(),
                           nnz);
      }
      hipLaunchKernelGGL((row_sort_kernel<T, index_t>), dim3(m), dim3(64), 0,
                         cur_stream(), indptr.data_ptr<int64_t>(),
                         out_idx.data_ptr<index_t>(), out_vals.data_ptr<T>(),
                         m, flags.data_ptr<int>());
    });
  });
}

// -- dense -> CSR (two-phase) -------------------------------------------
// Reference parity: DENSE_TO_CSR_NNZ / DENSE_TO_CSR (dense_to_csr.cu) as a
// real HIP kernel (VERDICT r1 partial): one wave per row; per-64-column
// chunk a ballot mask gives each nonzero its ordered slot (popcount of
// lower lanes) — coalesced reads, ordered compact writes, no atomics.
namespace {

template <typename T>
__device__ __forceinline__ bool nz_of(T v) { return v != T(0); }
template <typename T>
__device__ __forceinline__ bool nz_of(c10::complex<T> v) {
  return v.real() != T(0) || v.imag() != T(0);
}

template <typename T, typename index_t>
__global__ __launch_bounds__(64) void dense_to_csr_kernel(
    const T* __restrict__ D, int64_t m, int64_t n, int64_t stride,
    const int64_t* __restrict__ indptr /*null in count phase*/,
    int64_t* __restrict__ nnz_per_row, index_t* __restrict__ indices,
    T* __restrict__ vals) {
  const int64_t row = blockIdx.x;
  if (row >= m) return;
  const int lane = threadIdx.x;
  int64_t base = indptr ? indptr[row] : 0;
  int64_t count = 0;
  for (int64_t c0 = 0; c0 < n; c0 += 64) {
    const int64_t c = c0 + lane;
    T v = (c < n) ? D[row * stride + c] : T(0);
    const bool nz = (c < n) && nz_of(v);
    const uint64_t mask = __ballot(nz);
    if (indptr && nz) {
      const int off = __popcll(mask & ((1ull << lane) - 1ull));
      indices[base + off] = (index_t)c;
      vals[base + off] = v;
    }
    const int chunk = __popcll(mask);
    base += chunk;
    count += chunk;
  }
  if (!indptr && lane == 0) nnz_per_row[row] = count;
}

}  // namespace

void dense_to_csr_hip(at::Tensor D, at::Tensor indptr_or_counts,
                      at::Tensor indices, at::Tensor vals, bool fill) {
  int64_t m = D.size(0), n = D.size(1), stride = D.stride(0);
  if (m == 0) return;
  TORCH_CHECK(D.stride(1) == 1, "dense_to_csr: row-major input");
  DISPATCH_VALUES(D.scalar_type(), "dense_to_csr", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(fill ? indices.scalar_type() : at::kLong,
                   "dense_to_csr_idx", [&] {
      hipLaunchKernelGGL(
          (dense_to_csr_kernel<T, index_t>), dim3(m), dim3(64), 0,
          cur_stream(), D.data_ptr<T>(), m, n, stride,
          fill ? indptr_or_counts.data_ptr<int64_t>() : nullptr,
          fill ? nullptr : indptr_or_counts.data_ptr<int64_t>(),
          fill ? indices.data_ptr<index_t>() : nullptr,
          fill ? vals.data_ptr<T>() : nullptr);
    });
  });
}
