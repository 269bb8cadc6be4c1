// CSR SpMV, nnz-split with LDS-staged products — MI355X (gfx950) native.
//
// Replaces reference CSR_SPMV_ROW_SPLIT (src/sparse/array/csr/spmv.cu:25-123,
// cuSPARSE-backed there).  Design for CDNA4 (measured on MI355X):
//  - each 256-thread block owns NNZ_PER_BLOCK consecutive nonzeros; values
//    and indices stream through 2-element vector loads (512B-1KB per wave
//    instruction, fully coalesced, read exactly once);
//  - ALL global loads (values, indices, x gathers, the block's indptr
//    segment) issue in one phase before the single barrier; the row-sum
//    phase then runs purely out of LDS — PMC showed the original
//    two-phase form 90% wave-parked because the sum phase had ~2 loads in
//    flight per wave (3.3 TB/s); this restructure lifts HBM pressure;
//  - XCD-aware block swizzle (common.h) keeps neighbouring nnz chunks on
//    one XCD's private L2 for x-gather reuse;
//  - rows cut by a block boundary produce one carry per block, combined by
//    a tiny fixup kernel (atomic-free main path);
//  - optional fused dot accumulates sum(p[r]*y[r]) per block (CG p·Ap);
//    the fixup kernel reduces the per-block partials.
#include <type_traits>

#include "common.h"

namespace {

constexpr int BLK = 256;
constexpr int VT = 8;  // nnz per thread (2 vector quads)
constexpr int64_t NNZ_PER_BLOCK = (int64_t)BLK * VT;  // 2048

template <typename U>
struct alignas(4 * sizeof(U) <= 16 ? 4 * sizeof(U) : 16) Quad {
  U v[4];
};

template <typename T, typename index_t, bool BETA_ZERO, bool FUSE_DOT>
__global__ __launch_bounds__(BLK) void spmv_kernel(
    const int64_t* __restrict__ indptr,  // m+1
    const index_t* __restrict__ indices,
    const T* __restrict__ vals,
    const T* __restrict__ x,  // window, indexed by (indices[p] - col_lo)
    T* __restrict__ y,
    int64_t m, int64_t nnz, int64_t col_lo, T beta,
    T* __restrict__ carry_val, int64_t* __restrict__ carry_row,
    const T* __restrict__ pvec,  // local p slab (rows align with y); FUSE_DOT
    T* __restrict__ dot_partial) {
  extern __shared__ char smem_raw[];
  T* prod = reinterpret_cast<T*>(smem_raw);  // NNZ_PER_BLOCK
  __shared__ __align__(16) char red_raw[BLK * sizeof(T)];
  T* red = reinterpret_cast<T*>(red_raw);

  const int64_t b = xcd_swizzle(blockIdx.x, gridDim.x);
  const int64_t s = b * NNZ_PER_BLOCK;
  const int64_t e = min(s + NNZ_PER_BLOCK, nnz);
  const int tid = threadIdx.x;

  // owned rows (redundant per-thread binary search; uniform -> broadcast)
  const int64_t ro0 = lb_i64(indptr, m, s);
  const int64_t ro1 = (e == nnz) ? m : lb_i64(indptr, m, e);

  // ---- phase 1: issue every global load ----------------------------------
  const bool full = (e - s) == NNZ_PER_BLOCK;
  if (full) {
    Quad<index_t> idx4[VT / 4];
    Quad<T> v4[VT / 4];
#pragma unroll
    for (int k = 0; k < VT / 4; ++k) {
      idx4[k] = *reinterpret_cast<const Quad<index_t>*>(
          &indices[s + tid * 4 + (int64_t)k * (4 * BLK)]);
    }
#pragma unroll
    for (int k = 0; k < VT / 4; ++k) {
      v4[k] = *reinterpret_cast<const Quad<T>*>(
          &vals[s + tid * 4 + (int64_t)k * (4 * BLK)]);
    }
    T xv[VT];
#pragma unroll
    for (int k = 0; k < VT / 4; ++k) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        xv[k * 4 + j] = x[(int64_t)idx4[k].v[j] - col_lo];
      }
    }
#pragma unroll
    for (int k = 0; k < VT / 4; ++k) {
      Quad<T> pr;
#pragma unroll
      for (int j = 0; j < 4; ++j) pr.v[j] = v4[k].v[j] * xv[k * 4 + j];
      *reinterpret_cast<Quad<T>*>(&prod[tid * 4 + k * (4 * BLK)]) = pr;
    }
  } else {
    for (int64_t i = s + tid; i < e; i += BLK) {
      prod[i - s] = vals[i] * x[(int64_t)indices[i] - col_lo];
    }
  }
  __syncthreads();

  // ---- phase 2: per-thread row sums out of LDS ---------------------------
  T dacc = ZeroOf<T>::value();
  for (int64_t r = ro0 + tid; r < ro1; r += BLK) {
    const int64_t rs = indptr[r];
    const int64_t re = min(indptr[r + 1], e);
    T acc = ZeroOf<T>::value();
    for (int64_t p = rs; p < re; ++p) acc += prod[p - s];
    if (BETA_ZERO) {
      y[r] = acc;
    } else {
      acc = acc + beta * y[r];
      y[r] = acc;
    }
    if (FUSE_DOT) dacc += acc * pvec[r];
  }

  // continuation carry: items [s, cend) belong to row ro0-1
  bool has_carry = false;
  if (ro0 > 0) {
    int64_t cend;
    if (ro0 < m) {
      cend = min(indptr[ro0], e);
    } else {
      cend = e;
    }
    if (cend > s) {
      has_carry = true;
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
    }
  }
  if (!has_carry && tid == 0) carry_row[b] = -1;

  if (FUSE_DOT) {
    __syncthreads();
    red[tid] = dacc;
    __syncthreads();
    for (int w = BLK / 2; w > 0; w >>= 1) {
      if (tid < w) red[tid] += red[tid + w];
      __syncthreads();
    }
    // per-block partial; one atomic per address would serialize at 163k
    // blocks, so the fixup kernel reduces these (nblocks/256 atomics).
    if (tid == 0) dot_partial[b] = red[0];
  }
}

template <typename T, bool FUSE_DOT>
__global__ void carry_fixup_kernel(const T* __restrict__ carry_val,
                                   const int64_t* __restrict__ carry_row,
                                   T* __restrict__ y, int64_t nblocks,
                                   const T* __restrict__ pvec,
                                   const T* __restrict__ dot_partial,
                                   T* __restrict__ dot_out) {
  __shared__ __align__(16) char red_raw[256 * sizeof(T)];
  T* red = reinterpret_cast<T*>(red_raw);
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  T dacc = ZeroOf<T>::value();
  if (i < nblocks) {
    int64_t r = carry_row[i];
    if (r >= 0) {
      T c = carry_val[i];
      atomic_add_any(&y[r], c);
      if (FUSE_DOT) dacc = c * pvec[r];
    }
    if (FUSE_DOT) dacc += dot_partial[i];
  }
  if (FUSE_DOT) {
    red[threadIdx.x] = dacc;
    __syncthreads();
    for (int w = 128; w > 0; w >>= 1) {
      if ((int)threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
      __syncthreads();
    }
    if (threadIdx.x == 0) atomic_add_any(dot_out, red[0]);
  }
}

template <typename T>
__global__ void scale_kernel(T* y, int64_t n, T beta) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) y[i] = beta * y[i];
}

}  // namespace

static void spmv_impl(at::Tensor indptr, at::Tensor indices, at::Tensor values,
                      at::Tensor x, at::Tensor y, int64_t col_lo, double beta,
                      const c10::optional<at::Tensor>& pvec,
                      const c10::optional<at::Tensor>& dot_out) {
  const int64_t m = indptr.numel() - 1;
  const int64_t nnz = values.numel();
  auto stream = cur_stream();
  const bool fuse_dot = pvec.has_value();
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
    at::Tensor dot_partial;
    if (fuse_dot) dot_partial = at::empty({nblocks}, values.options());
    size_t smem = NNZ_PER_BLOCK * sizeof(T);
    const T* pp = fuse_dot ? pvec->data_ptr<T>() : nullptr;
    T* dpart = fuse_dot ? dot_partial.data_ptr<T>() : nullptr;
    T* dp = fuse_dot ? dot_out->data_ptr<T>() : nullptr;
    DISPATCH_INDEX(indices.scalar_type(), "spmv_idx", [&] {
      auto launch = [&](auto kern) {
        hipLaunchKernelGGL(kern, dim3(nblocks), dim3(BLK), smem, stream,
                           indptr.data_ptr<int64_t>(), indices.data_ptr<index_t>(),
                           values.data_ptr<T>(), x.data_ptr<T>(), y.data_ptr<T>(),
                           m, nnz, col_lo, betav, carry_val.data_ptr<T>(),
                           carry_row.data_ptr<int64_t>(), pp, dpart);
      };
      if (beta == 0.0 && !fuse_dot) launch(spmv_kernel<T, index_t, true, false>);
      else if (beta == 0.0 && fuse_dot) launch(spmv_kernel<T, index_t, true, true>);
      else if (!fuse_dot) launch(spmv_kernel<T, index_t, false, false>);
      else launch(spmv_kernel<T, index_t, false, true>);
    });
    if (fuse_dot) {
      hipLaunchKernelGGL((carry_fixup_kernel<T, true>), dim3((nblocks + 255) / 256),
                         dim3(256), 0, stream, carry_val.data_ptr<T>(),
                         carry_row.data_ptr<int64_t>(), y.data_ptr<T>(), nblocks,
                         pp, dpart, dp);
    } else {
      hipLaunchKernelGGL((carry_fixup_kernel<T, false>), dim3((nblocks + 255) / 256),
                         dim3(256), 0, stream, carry_val.data_ptr<T>(),
                         carry_row.data_ptr<int64_t>(), y.data_ptr<T>(), nblocks,
                         pp, dpart, dp);
    }
  });
}

void spmv_hip(at::Tensor indptr, at::Tensor indices, at::Tensor values,
              at::Tensor x, at::Tensor y, int64_t col_lo, double beta) {
  spmv_impl(indptr, indices, values, x, y, col_lo, beta, c10::nullopt,
            c10::nullopt);
}

// fused q = A p ; dot_out += sum(p_local * q_local)
void spmv_dot_hip(at::Tensor indptr, at::Tensor indices, at::Tensor values,
                  at::Tensor x, at::Tensor y, at::Tensor pvec,
                  at::Tensor dot_out, int64_t col_lo) {
  spmv_impl(indptr, indices, values, x, y, col_lo, 0.0, pvec, dot_out);
}

// ---------------------------------------------------------------------------
// Padded-ELL fast path (column-major).  For row-uniform matrices (FD stencils,
// banded operators — the headline BASELINE workloads) the ELL mirror makes
// every value/index load perfectly coalesced with no indptr reads and no row
// search: measured 1.9x over the nnz-split CSR kernel on 5-pt Poisson fp64
// (0.94 ms vs 1.78 ms at nx=8192, ~6 TB/s effective; tools/spmv_bench.hip).
// The mirror is built once per matrix structure and cached on the csr_array.
namespace {

template <typename T, typename index_t>
__global__ void build_ell_kernel(const int64_t* __restrict__ indptr,
                                 const index_t* __restrict__ indices,
                                 const T* __restrict__ vals,
                                 index_t* __restrict__ eidx,
                                 T* __restrict__ evals, int64_t m, int64_t mp,
                                 int W, index_t pad_idx) {
  int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (r >= mp) return;
  if (r >= m) {
    for (int k = 0; k < W; ++k) {
      eidx[(int64_t)k * mp + r] = pad_idx;
      evals[(int64_t)k * mp + r] = ZeroOf<T>::value();
    }
    return;
  }
  const int64_t rs = indptr[r];
  const int64_t cnt = indptr[r + 1] - rs;
  const index_t pi = cnt > 0 ? indices[rs] : pad_idx;
  for (int k = 0; k < W; ++k) {
    if (k < cnt) {
      eidx[(int64_t)k * mp + r] = indices[rs + k];
      evals[(int64_t)k * mp + r] = vals[rs + k];
    } else {
      eidx[(int64_t)k * mp + r] = pi;
      evals[(int64_t)k * mp + r] = ZeroOf<T>::value();
    }
  }
}

// scatter the CSR values into the padded diagonal planes (dvals assumed
// pre-zeroed): one thread per ROW walking its short slice — replaces an
// nnz-scale torch index_put_ in the mirror build (indexFuncLargeIndex was
// ~79 ms/call on the 931M-nnz 3-D GMG fine level)
template <typename T, typename index_t>
__global__ void build_dia_kernel(const int64_t* __restrict__ indptr,
                                 const index_t* __restrict__ indices,
                                 const T* __restrict__ vals,
                                 const int64_t* __restrict__ offs,
                                 T* __restrict__ dvals, int64_t m, int64_t mp,
                                 int W, int64_t row0) {
  const int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (r >= m) return;
  const int64_t e = indptr[r + 1];
  for (int64_t p = indptr[r]; p < e; ++p) {
    const int64_t diag = (int64_t)indices[p] - (row0 + r);
    // offs sorted ascending, W <= 48: binary search
    int lo = 0, hi = W;
    while (lo < hi) {
      const int mid = (lo + hi) >> 1;
      if (offs[mid] < diag) lo = mid + 1; else hi = mid;
    }
    dvals[(int64_t)lo * mp + r] = vals[p];
  }
}

// nt_load (common.h): measured on the 16384^2 5-pt DIA SpMV:
// 2.580 -> 2.474 ms (tools/dia_nt_bench.hip); nt STORES measured slower,
// so outputs use plain stores.

// Window pair load [c0, c0+1] from a single piece (SINGLE mode), clamped
// at the edges.  Rows r0, r0+1 of one thread access consecutive columns on
// every diagonal, and the alignment parity (c0 & 1) is uniform per
// diagonal (r0 is even), so the interior fast path is one 16B load (even
// parity) or two scalars (odd) — halving the load-issue count that bounds
// the DIA kernels (profiled 4.6 TB/s issue-bound vs 6.0 achievable).
template <typename T>
__device__ __forceinline__ void wpair(const T* __restrict__ v, int64_t c0,
                                      int64_t wsize, T& o0, T& o1) {
  if (c0 >= 0 && c0 + 1 < wsize) {
    if ((c0 & 1) == 0) {
      struct alignas(2 * sizeof(T) <= 16 ? 2 * sizeof(T) : 16) TP { T a, b; };
      const TP t = *reinterpret_cast<const TP*>(&v[c0]);
      o0 = t.a;
      o1 = t.b;
    } else {
      o0 = v[c0];
      o1 = v[c0 + 1];
    }
  } else {
    o0 = v[min(max(c0, (int64_t)0), wsize - 1)];
    o1 = v[min(max(c0 + 1, (int64_t)0), wsize - 1)];
  }
}

// window-relative x lookup over (halo_lo | own slab | halo_hi) — the own
// piece is the rank's x slab used IN PLACE (no per-SpMV self-copy).
template <typename T>
__device__ __forceinline__ T xpiece(int64_t idx, const T* __restrict__ hlo,
                                    int64_t nlo, const T* __restrict__ own,
                                    int64_t nown, const T* __restrict__ hhi) {
  if (idx < nlo) return hlo[idx];
  idx -= nlo;
  if (idx < nown) return own[idx];
  return hhi[idx - nown];
}

template <typename T, typename index_t, bool FUSE_DOT, bool SINGLE>
__global__ __launch_bounds__(BLK) void ell_spmv_kernel(
    const index_t* __restrict__ eidx, const T* __restrict__ evals,
    const T* __restrict__ hlo, const T* __restrict__ own,
    const T* __restrict__ hhi, T* __restrict__ y,
    const T* __restrict__ pvec, T* __restrict__ dot_partial,
    int64_t m, int64_t mp, int W, int64_t col_lo, int64_t nlo, int64_t nown,
    int64_t rbase, int64_t rhi) {
  // [rbase, rhi): row sub-range (rbase even) — the interior/boundary split
  // that overlaps halo exchange with interior compute at ws>1 (same
  // contract as dia_spmv_kernel)
  __shared__ __align__(16) char red_raw[BLK * sizeof(T)];
  T* red = reinterpret_cast<T*>(red_raw);
  const int64_t t = (int64_t)blockIdx.x * BLK + threadIdx.x;
  const int64_t r0 = rbase + 2 * t;
  T a0 = ZeroOf<T>::value(), a1 = ZeroOf<T>::value();
  if (r0 < rhi) {
    for (int k = 0; k < W; ++k) {
      const int64_t base = (int64_t)k * mp + r0;
      // adjacent pair: one 2-element vector load per stream
      struct alignas(2 * sizeof(index_t) <= 16 ? 2 * sizeof(index_t) : 16) IP { index_t a, b; };
      struct alignas(2 * sizeof(T) <= 16 ? 2 * sizeof(T) : 16) TP { T a, b; };
      IP ii;
      ii.a = nt_load(&eidx[base]);
      ii.b = nt_load(&eidx[base + 1]);
      TP vv;
      vv.a = nt_load(&evals[base]);
      vv.b = nt_load(&evals[base + 1]);
      if (SINGLE) {
        // ws=1 fast case: the whole window is the own slab — no piece branch
        a0 += vv.a * own[(int64_t)ii.a - col_lo];
        a1 += vv.b * own[(int64_t)ii.b - col_lo];
      } else {
        a0 += vv.a * xpiece((int64_t)ii.a - col_lo, hlo, nlo, own, nown, hhi);
        a1 += vv.b * xpiece((int64_t)ii.b - col_lo, hlo, nlo, own, nown, hhi);
      }
    }
    if (r0 + 1 < min(m, rhi)) {
      struct alignas(2 * sizeof(T) <= 16 ? 2 * sizeof(T) : 16) TP { T a, b; };
      TP out{a0, a1};
      *reinterpret_cast<TP*>(&y[r0]) = out;
    } else if (r0 < m) {
      y[r0] = a0;
    }
  }
  if (FUSE_DOT) {
    T d = ZeroOf<T>::value();
    if (r0 < min(m, rhi)) d += a0 * pvec[r0];
    if (r0 + 1 < min(m, rhi)) d += a1 * pvec[r0 + 1];
    red[threadIdx.x] = d;
    __syncthreads();
    for (int w = BLK / 2; w > 0; w >>= 1) {
      if ((int)threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
      __syncthreads();
    }
    if (threadIdx.x == 0) dot_partial[blockIdx.x] = red[0];
  }
}

}  // namespace

void build_ell_hip(at::Tensor indptr, at::Tensor indices, at::Tensor values,
                   at::Tensor eidx, at::Tensor evals, int64_t W,
                   int64_t pad_idx) {
  const int64_t m = indptr.numel() - 1;
  const int64_t mp = evals.numel() / W;
  DISPATCH_VALUES(values.scalar_type(), "build_ell", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(indices.scalar_type(), "build_ell_idx", [&] {
      hipLaunchKernelGGL((build_ell_kernel<T, index_t>), dim3((mp + 255) / 256),
                         dim3(256), 0, cur_stream(), indptr.data_ptr<int64_t>(),
                         indices.data_ptr<index_t>(), values.data_ptr<T>(),
                         eidx.data_ptr<index_t>(), evals.data_ptr<T>(), m, mp,
                         (int)W, (index_t)pad_idx);
    });
  });
}

void ell_spmv_hip(at::Tensor eidx, at::Tensor evals, at::Tensor hlo,
                  at::Tensor own, at::Tensor hhi, at::Tensor y,
                  int64_t W, int64_t m, int64_t col_lo,
                  const c10::optional<at::Tensor>& pvec,
                  const c10::optional<at::Tensor>& dot_partial,
                  int64_t rbase, int64_t rhi) {
  const int64_t mp = evals.numel() / W;
  if (rhi < 0) rhi = mp;
  if (rhi <= rbase) return;
  const bool fuse = pvec.has_value();
  const int64_t nblocks = ((rhi - rbase) / 2 + BLK - 1) / BLK;
  const int64_t nlo = hlo.numel();
  const int64_t nown = own.numel();
  DISPATCH_VALUES(evals.scalar_type(), "ell_spmv", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(eidx.scalar_type(), "ell_spmv_idx", [&] {
      const T* hlo_p = nlo ? hlo.data_ptr<T>() : own.data_ptr<T>();
      const T* hhi_p = hhi.numel() ? hhi.data_ptr<T>() : own.data_ptr<T>();
      const bool single = (nlo == 0 && hhi.numel() == 0);
      auto launch = [&](auto kern, const T* pv, T* dp) {
        hipLaunchKernelGGL(kern, dim3(nblocks), dim3(BLK), 0, cur_stream(),
                           eidx.data_ptr<index_t>(), evals.data_ptr<T>(),
                           hlo_p, own.data_ptr<T>(), hhi_p, y.data_ptr<T>(),
                           pv, dp, m, mp, (int)W, col_lo, nlo, nown,
                           rbase, rhi);
      };
      if (fuse && single)
        launch(ell_spmv_kernel<T, index_t, true, true>, pvec->data_ptr<T>(),
               dot_partial->data_ptr<T>());
      else if (fuse)
        launch(ell_spmv_kernel<T, index_t, true, false>, pvec->data_ptr<T>(),
               dot_partial->data_ptr<T>());
      else if (single)
        launch(ell_spmv_kernel<T, index_t, false, true>, nullptr, nullptr);
      else
        launch(ell_spmv_kernel<T, index_t, false, false>, nullptr, nullptr);
    });
  });
}

void build_dia_hip(at::Tensor indptr, at::Tensor indices, at::Tensor values,
                   at::Tensor offs, at::Tensor dvals, int64_t W,
                   int64_t row0) {
  const int64_t m = indptr.numel() - 1;
  const int64_t mp = dvals.numel() / W;
  if (m == 0) return;
  DISPATCH_VALUES(values.scalar_type(), "build_dia", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(indices.scalar_type(), "build_dia_idx", [&] {
      hipLaunchKernelGGL((build_dia_kernel<T, index_t>),
                         dim3((m + 255) / 256), dim3(256), 0, cur_stream(),
                         indptr.data_ptr<int64_t>(),
                         indices.data_ptr<index_t>(), values.data_ptr<T>(),
                         offs.data_ptr<int64_t>(), dvals.data_ptr<T>(), m, mp,
                         (int)W, row0);
    });
  });
}

void ell_spmv_plain_hip(at::Tensor eidx, at::Tensor evals, at::Tensor hlo,
                        at::Tensor own, at::Tensor hhi, at::Tensor y,
                        int64_t W, int64_t m, int64_t col_lo,
                        int64_t rbase, int64_t rhi) {
  ell_spmv_hip(eidx, evals, hlo, own, hhi, y, W, m, col_lo, c10::nullopt,
               c10::nullopt, rbase, rhi);
}

void ell_spmv_dot_hip(at::Tensor eidx, at::Tensor evals, at::Tensor hlo,
                      at::Tensor own, at::Tensor hhi, at::Tensor y,
                      at::Tensor pvec, at::Tensor dot_partial,
                      int64_t W, int64_t m, int64_t col_lo,
                      int64_t rbase, int64_t rhi) {
  ell_spmv_hip(eidx, evals, hlo, own, hhi, y, W, m, col_lo, pvec, dot_partial,
               rbase, rhi);
}

// ---------------------------------------------------------------------------
// Fused weighted-Jacobi sweep on the ELL mirror:
//   x_out[r] = x[r] + omega * dinv[r] * (b[r] - (A x)[r])
// One pass over A + 4 vector streams instead of spmv + 2 elementwise
// kernels (the GMG/AMG smoother, reference WeightedJacobi gmg.py:247-285).
namespace {

template <typename T, typename index_t, bool SINGLE>
__global__ __launch_bounds__(BLK) void ell_jacobi_kernel(
    const index_t* __restrict__ eidx, const T* __restrict__ evals,
    const T* __restrict__ hlo, const T* __restrict__ own,
    const T* __restrict__ hhi, const T* __restrict__ xloc,
    const T* __restrict__ b, const T* __restrict__ dinv,
    T* __restrict__ xout, int64_t m, int64_t mp, int W, int64_t col_lo,
    int64_t nlo, int64_t nown, T omega) {
  const int64_t t = (int64_t)blockIdx.x * BLK + threadIdx.x;
  const int64_t r0 = 2 * t;
  if (r0 >= mp) return;
  T a0 = ZeroOf<T>::value(), a1 = ZeroOf<T>::value();
  for (int k = 0; k < W; ++k) {
    const int64_t base = (int64_t)k * mp + r0;
    struct alignas(2 * sizeof(index_t) <= 16 ? 2 * sizeof(index_t) : 16) IP { index_t a, b; };
    struct alignas(2 * sizeof(T) <= 16 ? 2 * sizeof(T) : 16) TP { T a, b; };
    IP ii;
    ii.a = nt_load(&eidx[base]);
    ii.b = nt_load(&eidx[base + 1]);
    TP vv;
    vv.a = nt_load(&evals[base]);
    vv.b = nt_load(&evals[base + 1]);
    if (SINGLE) {
      a0 += vv.a * own[(int64_t)ii.a - col_lo];
      a1 += vv.b * own[(int64_t)ii.b - col_lo];
    } else {
      a0 += vv.a * xpiece((int64_t)ii.a - col_lo, hlo, nlo, own, nown, hhi);
      a1 += vv.b * xpiece((int64_t)ii.b - col_lo, hlo, nlo, own, nown, hhi);
    }
  }
  if (r0 + 1 < m) {
    struct alignas(2 * sizeof(T) <= 16 ? 2 * sizeof(T) : 16) TP { T a, b; };
    const TP xv = *reinterpret_cast<const TP*>(&xloc[r0]);
    const TP bv = *reinterpret_cast<const TP*>(&b[r0]);
    const TP dv = *reinterpret_cast<const TP*>(&dinv[r0]);
    TP out{xv.a + omega * dv.a * (bv.a - a0),
           xv.b + omega * dv.b * (bv.b - a1)};
    *reinterpret_cast<TP*>(&xout[r0]) = out;
  } else if (r0 < m) {
    xout[r0] = xloc[r0] + omega * dinv[r0] * (b[r0] - a0);
  }
}

}  // namespace

void ell_jacobi_hip(at::Tensor eidx, at::Tensor evals, at::Tensor hlo,
                    at::Tensor own, at::Tensor hhi, at::Tensor xloc,
                    at::Tensor b, at::Tensor dinv, at::Tensor xout,
                    int64_t W, int64_t m, int64_t col_lo, double omega) {
  const int64_t mp = evals.numel() / W;
  const int64_t nblocks = (mp / 2 + BLK - 1) / BLK;
  const int64_t nlo = hlo.numel();
  const int64_t nown = own.numel();
  DISPATCH_VALUES(evals.scalar_type(), "ell_jacobi", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(eidx.scalar_type(), "ell_jacobi_idx", [&] {
      const T* hlo_p = nlo ? hlo.data_ptr<T>() : own.data_ptr<T>();
      const T* hhi_p = hhi.numel() ? hhi.data_ptr<T>() : own.data_ptr<T>();
      const bool single = (nlo == 0 && hhi.numel() == 0);
      auto launch = [&](auto kern) {
        hipLaunchKernelGGL(kern, dim3(nblocks), dim3(BLK), 0, cur_stream(),
                           eidx.data_ptr<index_t>(), evals.data_ptr<T>(),
                           hlo_p, own.data_ptr<T>(), hhi_p, xloc.data_ptr<T>(),
                           b.data_ptr<T>(), dinv.data_ptr<T>(),
                           xout.data_ptr<T>(), m, mp, (int)W, col_lo, nlo,
                           nown, static_cast<T>(omega));
      };
      if (single) launch(ell_jacobi_kernel<T, index_t, true>);
      else launch(ell_jacobi_kernel<T, index_t, false>);
    });
  });
}

// ---------------------------------------------------------------------------
// Row-per-thread CSR SpMV: for short-row matrices that the ELL heuristic
// rejects (padding blowup), a plain thread-per-row loop measured 3.7 TB/s
// vs the nnz-split kernel's 3.2 on short rows (tools/spmv_bench.hip v1/v2);
// the nnz-split kernel remains the choice for long/skewed rows.
namespace {

template <typename T, typename index_t>
__global__ __launch_bounds__(BLK) void csr_row_spmv_kernel(
    const int64_t* __restrict__ indptr, const index_t* __restrict__ indices,
    const T* __restrict__ vals, const T* __restrict__ x, T* __restrict__ y,
    int64_t m, int64_t col_lo) {
  const int64_t b = xcd_swizzle(blockIdx.x, gridDim.x);
  const int64_t r = b * BLK + threadIdx.x;
  if (r >= m) return;
  const int64_t e = indptr[r + 1];
  T acc = ZeroOf<T>::value();
  for (int64_t p = indptr[r]; p < e; ++p) {
    acc += vals[p] * x[(int64_t)indices[p] - col_lo];
  }
  y[r] = acc;
}

}  // namespace

void csr_row_spmv_hip(at::Tensor indptr, at::Tensor indices, at::Tensor values,
                      at::Tensor x, at::Tensor y, int64_t col_lo) {
  const int64_t m = indptr.numel() - 1;
  if (m == 0) return;
  DISPATCH_VALUES(values.scalar_type(), "csr_row_spmv", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(indices.scalar_type(), "csr_row_spmv_idx", [&] {
      hipLaunchKernelGGL((csr_row_spmv_kernel<T, index_t>),
                         dim3((m + BLK - 1) / BLK), dim3(BLK), 0, cur_stream(),
                         indptr.data_ptr<int64_t>(), indices.data_ptr<index_t>(),
                         values.data_ptr<T>(), x.data_ptr<T>(), y.data_ptr<T>(),
                         m, col_lo);
    });
  });
}

// ---------------------------------------------------------------------------
// Device-DIA fast path: when the matrix has few distinct diagonals
// (col - row values), store ONLY the padded value planes column-major plus
// the W offsets — the index stream disappears (12 -> 8 B/nnz for
// fp64+int32).  Invalid/padded entries hold 0 and their x loads are clamped
// into the window.  Reference context: the dia format (sparse/dia.py) is a
// host/conversion format there; here it is the SpMV execution format for
// banded operators (Poisson, dot_microbenchmark).
namespace {

template <typename T, bool FUSE_DOT, bool SINGLE, bool RESID = false>
__global__ __launch_bounds__(BLK) void dia_spmv_kernel(
    const T* __restrict__ dvals,  // (W, mp) column-major planes
    const int64_t* __restrict__ offs,  // W diagonal offsets
    const T* __restrict__ hlo, const T* __restrict__ own,
    const T* __restrict__ hhi, T* __restrict__ y,
    const T* __restrict__ pvec, T* __restrict__ dot_partial,
    int64_t m, int64_t mp, int W, int64_t col_lo, int64_t row0,
    int64_t nlo, int64_t nown, int64_t wsize, int64_t rbase, int64_t rhi) {
  // [rbase, rhi): row sub-range (rbase even) — the interior/boundary split
  // that overlaps halo exchange with interior compute at ws>1
  __shared__ __align__(16) char red_raw[BLK * sizeof(T)];
  T* red = reinterpret_cast<T*>(red_raw);
  const int64_t t = (int64_t)blockIdx.x * BLK + threadIdx.x;
  const int64_t r0 = rbase + 2 * t;
  T a0 = ZeroOf<T>::value(), a1 = ZeroOf<T>::value();
  if (r0 < rhi) {
    for (int k = 0; k < W; ++k) {
      const int64_t base = (int64_t)k * mp + r0;
      struct alignas(2 * sizeof(T) <= 16 ? 2 * sizeof(T) : 16) TP { T a, b; };
      TP vv;
      vv.a = nt_load(&dvals[base]);
      vv.b = nt_load(&dvals[base + 1]);
      // window-relative column; padded entries are 0 so a clamped load is safe
      const int64_t c0 = row0 + r0 + offs[k] - col_lo;
      if (SINGLE) {
        T x0, x1;
        wpair(own, c0, wsize, x0, x1);
        a0 += vv.a * x0;
        a1 += vv.b * x1;
      } else {
        const int64_t i0 = min(max(c0, (int64_t)0), wsize - 1);
        const int64_t i1 = min(max(c0 + 1, (int64_t)0), wsize - 1);
        a0 += vv.a * xpiece(i0, hlo, nlo, own, nown, hhi);
        a1 += vv.b * xpiece(i1, hlo, nlo, own, nown, hhi);
      }
    }
    if (r0 + 1 < min(m, rhi)) {
      struct alignas(2 * sizeof(T) <= 16 ? 2 * sizeof(T) : 16) TP { T a, b; };
      TP out{a0, a1};
      if (RESID) {
        // y = b - A x (pvec carries b): the V-cycle residual fused into
        // the SpMV — saves the separate 3-pass pointwise kernel
        const TP bv = *reinterpret_cast<const TP*>(&pvec[r0]);
        out.a = bv.a - a0;
        out.b = bv.b - a1;
      }
      *reinterpret_cast<TP*>(&y[r0]) = out;
    } else if (r0 < m) {
      y[r0] = RESID ? (pvec[r0] - a0) : a0;
    }
  }
  if (FUSE_DOT) {
    T d = ZeroOf<T>::value();
    if (r0 < min(m, rhi)) d += a0 * pvec[r0];
    if (r0 + 1 < min(m, rhi)) d += a1 * pvec[r0 + 1];
    red[threadIdx.x] = d;
    __syncthreads();
    for (int w = BLK / 2; w > 0; w >>= 1) {
      if ((int)threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
      __syncthreads();
    }
    if (threadIdx.x == 0) dot_partial[blockIdx.x] = red[0];
  }
}

template <typename T, bool SINGLE>
__global__ __launch_bounds__(BLK) void dia_jacobi_kernel(
    const T* __restrict__ dvals, const int64_t* __restrict__ offs,
    const T* __restrict__ hlo, const T* __restrict__ own,
    const T* __restrict__ hhi, const T* __restrict__ xloc,
    const T* __restrict__ b, const T* __restrict__ dinv,
    T* __restrict__ xout, int64_t m, int64_t mp, int W, int64_t col_lo,
    int64_t row0, int64_t nlo, int64_t nown, int64_t wsize, T omega,
    int64_t rbase, int64_t rhi) {
  const int64_t t = (int64_t)blockIdx.x * BLK + threadIdx.x;
  const int64_t r0 = rbase + 2 * t;
  if (r0 >= rhi) return;
  T a0 = ZeroOf<T>::value(), a1 = ZeroOf<T>::value();
  for (int k = 0; k < W; ++k) {
    const int64_t base = (int64_t)k * mp + r0;
    struct alignas(2 * sizeof(T) <= 16 ? 2 * sizeof(T) : 16) TP { T a, b; };
    TP vv;
    vv.a = nt_load(&dvals[base]);
    vv.b = nt_load(&dvals[base + 1]);
    const int64_t c0 = row0 + r0 + offs[k] - col_lo;
    if (SINGLE) {
      T x0, x1;
      wpair(own, c0, wsize, x0, x1);
      a0 += vv.a * x0;
      a1 += vv.b * x1;
    } else {
      const int64_t i0 = min(max(c0, (int64_t)0), wsize - 1);
      const int64_t i1 = min(max(c0 + 1, (int64_t)0), wsize - 1);
      a0 += vv.a * xpiece(i0, hlo, nlo, own, nown, hhi);
      a1 += vv.b * xpiece(i1, hlo, nlo, own, nown, hhi);
    }
  }
  if (r0 + 1 < min(m, rhi)) {
    struct alignas(2 * sizeof(T) <= 16 ? 2 * sizeof(T) : 16) TP { T a, b; };
    const TP xv = *reinterpret_cast<const TP*>(&xloc[r0]);
    const TP bv = *reinterpret_cast<const TP*>(&b[r0]);
    const TP dv = *reinterpret_cast<const TP*>(&dinv[r0]);
    TP out{xv.a + omega * dv.a * (bv.a - a0),
           xv.b + omega * dv.b * (bv.b - a1)};
    *reinterpret_cast<TP*>(&xout[r0]) = out;
  } else if (r0 < m) {
    xout[r0] = xloc[r0] + omega * dinv[r0] * (b[r0] - a0);
  }
}

// Fully-fused CG K1 on the DIA mirror (two-kernel CG iteration):
//   p_new = r + beta * p_old   (beta = *beta_num / *beta_den, device scalars)
//   q     = A p_new
//   pq    = p_new . q          (per-block partials)
// p_new at neighbor columns is recomputed on the fly from the r / p_old
// windows (deterministic IEEE => identical to the stored value), so the
// separate p-update pass over 3 vector streams disappears.  p_old and
// p_new MUST be distinct buffers (double-buffered by the caller).
// Reference context: this is the MI355X fusion of the reference CG loop's
// AXPBY + SpMV + dot task chain (linalg.py:499-565).
template <typename T, bool SINGLE>
__global__ __launch_bounds__(BLK) void dia_spmv_bpdot_kernel(
    const T* __restrict__ dvals, const int64_t* __restrict__ offs,
    const T* __restrict__ r_hlo, const T* __restrict__ r_own,
    const T* __restrict__ r_hhi, const T* __restrict__ p_hlo,
    const T* __restrict__ p_own, const T* __restrict__ p_hhi,
    T* __restrict__ pnew, T* __restrict__ q,
    const T* __restrict__ beta_num, const T* __restrict__ beta_den,
    T* __restrict__ dot_partial, int64_t m, int64_t mp, int W,
    int64_t col_lo, int64_t row0, int64_t nlo, int64_t nown, int64_t wsize,
    int64_t own_off) {
  __shared__ __align__(16) char red_raw[BLK * sizeof(T)];
  T* red = reinterpret_cast<T*>(red_raw);
  const T beta = (*beta_num) / (*beta_den);
  const int64_t t = (int64_t)blockIdx.x * BLK + threadIdx.x;
  const int64_t r0 = 2 * t;
  T a0 = ZeroOf<T>::value(), a1 = ZeroOf<T>::value();
  T p0 = ZeroOf<T>::value(), p1 = ZeroOf<T>::value();
  if (r0 < mp) {
    for (int k = 0; k < W; ++k) {
      const int64_t base = (int64_t)k * mp + r0;
      T va = nt_load(&dvals[base]);
      T vb = nt_load(&dvals[base + 1]);
      const int64_t c0 = row0 + r0 + offs[k] - col_lo;
      T x0, x1;
      if (SINGLE) {
        T r0v, r1v, p0v, p1v;
        wpair(r_own, c0, wsize, r0v, r1v);
        wpair(p_own, c0, wsize, p0v, p1v);
        x0 = r0v + beta * p0v;
        x1 = r1v + beta * p1v;
      } else {
        const int64_t i0 = min(max(c0, (int64_t)0), wsize - 1);
        const int64_t i1 = min(max(c0 + 1, (int64_t)0), wsize - 1);
        x0 = xpiece(i0, r_hlo, nlo, r_own, nown, r_hhi)
             + beta * xpiece(i0, p_hlo, nlo, p_own, nown, p_hhi);
        x1 = xpiece(i1, r_hlo, nlo, r_own, nown, r_hhi)
             + beta * xpiece(i1, p_hlo, nlo, p_own, nown, p_hhi);
      }
      a0 += va * x0;
      a1 += vb * x1;
    }
    // own-row p_new (same formula as the window recompute => identical fp)
    const int64_t w0 = own_off + r0;
    if (SINGLE) {
      T r0v, r1v, p0v, p1v;
      wpair(r_own, w0, wsize, r0v, r1v);
      wpair(p_own, w0, wsize, p0v, p1v);
      p0 = r0v + beta * p0v;
      if (r0 + 1 < m) p1 = r1v + beta * p1v;
    } else {
      p0 = xpiece(w0, r_hlo, nlo, r_own, nown, r_hhi)
           + beta * xpiece(w0, p_hlo, nlo, p_own, nown, p_hhi);
      if (r0 + 1 < m)
        p1 = xpiece(w0 + 1, r_hlo, nlo, r_own, nown, r_hhi)
             + beta * xpiece(w0 + 1, p_hlo, nlo, p_own, nown, p_hhi);
    }
    if (r0 + 1 < m) {
      struct alignas(2 * sizeof(T) <= 16 ? 2 * sizeof(T) : 16) TP { T a, b; };
      TP po{p0, p1};
      *reinterpret_cast<TP*>(&pnew[r0]) = po;
      TP qo{a0, a1};
      *reinterpret_cast<TP*>(&q[r0]) = qo;
    } else if (r0 < m) {
      pnew[r0] = p0;
      q[r0] = a0;
    }
  }
  T d = ZeroOf<T>::value();
  if (r0 < m) d += a0 * p0;
  if (r0 + 1 < m) d += a1 * p1;
  red[threadIdx.x] = d;
  __syncthreads();
  for (int w = BLK / 2; w > 0; w >>= 1) {
    if ((int)threadIdx.x < w) red[threadIdx.x] += red[threadIdx.x + w];
    __syncthreads();
  }
  if (threadIdx.x == 0) dot_partial[blockIdx.x] = red[0];
}

}  // namespace

void dia_spmv_bpdot_hip(at::Tensor dvals, at::Tensor offs, at::Tensor r_hlo,
                        at::Tensor r_own, at::Tensor r_hhi, at::Tensor p_hlo,
                        at::Tensor p_own, at::Tensor p_hhi, at::Tensor pnew,
                        at::Tensor q, at::Tensor beta_num, at::Tensor beta_den,
                        at::Tensor dot_partial, int64_t W, int64_t m,
                        int64_t col_lo, int64_t row0, int64_t wsize) {
  const int64_t mp = dvals.numel() / W;
  const int64_t nblocks = (mp / 2 + BLK - 1) / BLK;
  const int64_t nlo = r_hlo.numel();
  const int64_t nown = r_own.numel();
  TORCH_CHECK(pnew.data_ptr() != p_own.data_ptr(),
              "dia_spmv_bpdot: pnew must not alias p_own (double-buffer)");
  TORCH_CHECK(dot_partial.numel() >= nblocks, "dia_spmv_bpdot: partial small");
  // own_off: window index of this rank's first own row (row0 - col_lo)
  const int64_t own_off_w = row0 - col_lo;
  // in SINGLE mode indices are into the own slab directly
  const bool single = (nlo == 0 && r_hhi.numel() == 0);
  DISPATCH_VALUES(dvals.scalar_type(), "dia_spmv_bpdot", [&] {
    using T = scalar_t;
    const T* rhlo_p = nlo ? r_hlo.data_ptr<T>() : r_own.data_ptr<T>();
    const T* rhhi_p = r_hhi.numel() ? r_hhi.data_ptr<T>() : r_own.data_ptr<T>();
    const T* phlo_p = nlo ? p_hlo.data_ptr<T>() : p_own.data_ptr<T>();
    const T* phhi_p = p_hhi.numel() ? p_hhi.data_ptr<T>() : p_own.data_ptr<T>();
    auto launch = [&](auto kern) {
      hipLaunchKernelGGL(kern, dim3(nblocks), dim3(BLK), 0, cur_stream(),
                         dvals.data_ptr<T>(), offs.data_ptr<int64_t>(),
                         rhlo_p, r_own.data_ptr<T>(), rhhi_p, phlo_p,
                         p_own.data_ptr<T>(), phhi_p, pnew.data_ptr<T>(),
                         q.data_ptr<T>(), beta_num.data_ptr<T>(),
                         beta_den.data_ptr<T>(), dot_partial.data_ptr<T>(),
                         m, mp, (int)W, col_lo, row0, nlo, nown, wsize,
                         own_off_w);
    };
    if (single) launch(dia_spmv_bpdot_kernel<T, true>);
    else launch(dia_spmv_bpdot_kernel<T, false>);
  });
}

void dia_spmv_hip(at::Tensor dvals, at::Tensor offs, at::Tensor hlo,
                  at::Tensor own, at::Tensor hhi, at::Tensor y,
                  int64_t W, int64_t m, int64_t col_lo, int64_t row0,
                  int64_t wsize,
                  const c10::optional<at::Tensor>& pvec,
                  const c10::optional<at::Tensor>& dot_partial,
                  int64_t rbase, int64_t rhi) {
  const int64_t mp = dvals.numel() / W;
  if (rhi < 0) rhi = mp;
  if (rhi <= rbase) return;
  const bool fuse = pvec.has_value();
  const int64_t nblocks = ((rhi - rbase) / 2 + BLK) / BLK;
  const int64_t nlo = hlo.numel();
  const int64_t nown = own.numel();
  DISPATCH_VALUES(dvals.scalar_type(), "dia_spmv", [&] {
    using T = scalar_t;
    const T* hlo_p = nlo ? hlo.data_ptr<T>() : own.data_ptr<T>();
    const T* hhi_p = hhi.numel() ? hhi.data_ptr<T>() : own.data_ptr<T>();
    const bool single = (nlo == 0 && hhi.numel() == 0);
    auto launch = [&](auto kern, const T* pv, T* dp) {
      hipLaunchKernelGGL(kern, dim3(nblocks), dim3(BLK), 0, cur_stream(),
                         dvals.data_ptr<T>(), offs.data_ptr<int64_t>(), hlo_p,
                         own.data_ptr<T>(), hhi_p, y.data_ptr<T>(), pv, dp,
                         m, mp, (int)W, col_lo, row0, nlo, nown, wsize,
                         rbase, rhi);
    };
    if (fuse && single)
      launch(dia_spmv_kernel<T, true, true>, pvec->data_ptr<T>(),
             dot_partial->data_ptr<T>());
    else if (fuse)
      launch(dia_spmv_kernel<T, true, false>, pvec->data_ptr<T>(),
             dot_partial->data_ptr<T>());
    else if (single)
      launch(dia_spmv_kernel<T, false, true>, nullptr, nullptr);
    else
      launch(dia_spmv_kernel<T, false, false>, nullptr, nullptr);
  });
}

void dia_residual_hip(at::Tensor dvals, at::Tensor offs, at::Tensor hlo,
                      at::Tensor own, at::Tensor hhi, at::Tensor b,
                      at::Tensor y, int64_t W, int64_t m, int64_t col_lo,
                      int64_t row0, int64_t wsize, int64_t rbase,
                      int64_t rhi) {
  // y = b - A x: the fused V-cycle residual (pvec slot carries b)
  const int64_t mp = dvals.numel() / W;
  if (rhi < 0) rhi = mp;
  if (rhi <= rbase) return;
  const int64_t nblocks = ((rhi - rbase) / 2 + BLK) / BLK;
  const int64_t nlo = hlo.numel();
  const int64_t nown = own.numel();
  DISPATCH_VALUES(dvals.scalar_type(), "dia_residual", [&] {
    using T = scalar_t;
    const T* hlo_p = nlo ? hlo.data_ptr<T>() : own.data_ptr<T>();
    const T* hhi_p = hhi.numel() ? hhi.data_ptr<T>() : own.data_ptr<T>();
    const bool single = (nlo == 0 && hhi.numel() == 0);
    auto launch = [&](auto kern) {
      hipLaunchKernelGGL(kern, dim3(nblocks), dim3(BLK), 0, cur_stream(),
                         dvals.data_ptr<T>(), offs.data_ptr<int64_t>(), hlo_p,
                         own.data_ptr<T>(), hhi_p, y.data_ptr<T>(),
                         b.data_ptr<T>(), nullptr, m, mp, (int)W, col_lo,
                         row0, nlo, nown, wsize, rbase, rhi);
    };
    if (single) launch(dia_spmv_kernel<T, false, true, true>);
    else launch(dia_spmv_kernel<T, false, false, true>);
  });
}

void dia_spmv_plain_hip(at::Tensor dvals, at::Tensor offs, at::Tensor hlo,
                        at::Tensor own, at::Tensor hhi, at::Tensor y,
                        int64_t W, int64_t m, int64_t col_lo, int64_t row0,
                        int64_t wsize, int64_t rbase, int64_t rhi) {
  dia_spmv_hip(dvals, offs, hlo, own, hhi, y, W, m, col_lo, row0, wsize,
               c10::nullopt, c10::nullopt, rbase, rhi);
}

void dia_spmv_dot_hip(at::Tensor dvals, at::Tensor offs, at::Tensor hlo,
                      at::Tensor own, at::Tensor hhi, at::Tensor y,
                      at::Tensor pvec, at::Tensor dot_partial, int64_t W,
                      int64_t m, int64_t col_lo, int64_t row0, int64_t wsize,
                      int64_t rbase, int64_t rhi) {
  dia_spmv_hip(dvals, offs, hlo, own, hhi, y, W, m, col_lo, row0, wsize,
               pvec, dot_partial, rbase, rhi);
}

void dia_jacobi_hip(at::Tensor dvals, at::Tensor offs, at::Tensor hlo,
                    at::Tensor own, at::Tensor hhi, at::Tensor xloc,
                    at::Tensor b, at::Tensor dinv, at::Tensor xout,
                    int64_t W, int64_t m, int64_t col_lo, int64_t row0,
                    int64_t wsize, double omega, int64_t rbase, int64_t rhi) {
  const int64_t mp = dvals.numel() / W;
  if (rhi < 0) rhi = mp;
  if (rhi <= rbase) return;
  const int64_t nblocks = ((rhi - rbase) / 2 + BLK) / BLK;
  const int64_t nlo = hlo.numel();
  const int64_t nown = own.numel();
  DISPATCH_VALUES(dvals.scalar_type(), "dia_jacobi", [&] {
    using T = scalar_t;
    const T* hlo_p = nlo ? hlo.data_ptr<T>() : own.data_ptr<T>();
    const T* hhi_p = hhi.numel() ? hhi.data_ptr<T>() : own.data_ptr<T>();
    const bool single = (nlo == 0 && hhi.numel() == 0);
    auto launch = [&](auto kern) {
      hipLaunchKernelGGL(kern, dim3(nblocks), dim3(BLK), 0, cur_stream(),
                         dvals.data_ptr<T>(), offs.data_ptr<int64_t>(), hlo_p,
                         own.data_ptr<T>(), hhi_p, xloc.data_ptr<T>(),
                         b.data_ptr<T>(), dinv.data_ptr<T>(),
                         xout.data_ptr<T>(), m, mp, (int)W, col_lo, row0,
                         nlo, nown, wsize, static_cast<T>(omega), rbase, rhi);
    };
    if (single) launch(dia_jacobi_kernel<T, true>);
    else launch(dia_jacobi_kernel<T, false>);
  });
}
