// SpGEMM: two-phase Gustavson with per-wave LDS hash accumulators,
// size-binned.
//
// Reference parity: SPGEMM_CSR_CSR_CSR_GPU (spgemm_csr_csr_csr.cu:33-272,
// cuSPARSE-backed there) and the CPU Gustavson semantics
// (spgemm_csr_csr_csr.cc:27-85).  MI355X design: one wave per C row, an LDS
// open-addressing hash (keys + fp accumulators) per wave.  Rows are BINNED
// by their product upper bound (sum of B-row sizes) into hash sizes
// 64/256/1024 so short stencil rows don't pay a 1024-entry table init;
// rows with ub > 512 fall to the wrapper's vectorized expand-sort-reduce.
// Output columns are compacted unsorted and sorted per-row by the wrapper.
#include "common.h"

namespace {

constexpr int WAVES_PER_BLOCK = 4;  // 256 threads

__device__ __forceinline__ uint32_t hash_mul(int64_t c) {
  return (uint32_t)((uint64_t)c * 2654435761u);
}

template <typename index_t, int HASH>
__global__ __launch_bounds__(256) void spgemm_nnz_kernel(
    const int64_t* __restrict__ aip, const index_t* __restrict__ aix,
    const int64_t* __restrict__ bip, const index_t* __restrict__ bix,
    const int64_t* __restrict__ rowlist, int64_t nrows,
    int64_t* __restrict__ nnz_out, int64_t a_col_lo) {
  __shared__ int64_t keys[WAVES_PER_BLOCK][HASH];
  __shared__ int counts[WAVES_PER_BLOCK];
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int64_t li = (int64_t)blockIdx.x * WAVES_PER_BLOCK + wid;
  if (li >= nrows) return;
  const int64_t r = rowlist[li];
  for (int i = lane; i < HASH; i += WAVE) keys[wid][i] = -1;
  if (lane == 0) counts[wid] = 0;
  __builtin_amdgcn_wave_barrier();
  __threadfence_block();
  const int64_t as = aip[r], ae = aip[r + 1];
  for (int64_t p = as + lane; p < ae; p += WAVE) {
    const int64_t brow = (int64_t)aix[p] - a_col_lo;
    const int64_t bs = bip[brow], be = bip[brow + 1];
    for (int64_t q = bs; q < be; ++q) {
      int64_t c = (int64_t)bix[q];
      uint32_t h = hash_mul(c) & (HASH - 1);
      while (true) {
        int64_t old = atomicCAS((unsigned long long*)&keys[wid][h],
                                (unsigned long long)(-1ll),
                                (unsigned long long)c);
        if (old == -1ll) { atomicAdd(&counts[wid], 1); break; }
        if (old == c) break;
        h = (h + 1) & (HASH - 1);
      }
    }
  }
  __builtin_amdgcn_wave_barrier();
  __threadfence_block();
  if (lane == 0) nnz_out[r] = (int64_t)counts[wid];
}

template <typename T, typename index_t, int HASH>
__global__ __launch_bounds__(256) void spgemm_compute_kernel(
    const int64_t* __restrict__ aip, const index_t* __restrict__ aix,
    const T* __restrict__ av, const int64_t* __restrict__ bip,
    const index_t* __restrict__ bix, const T* __restrict__ bv,
    const int64_t* __restrict__ rowlist, int64_t nrows,
    const int64_t* __restrict__ cip, index_t* __restrict__ cix,
    T* __restrict__ cv, int64_t a_col_lo) {
  __shared__ int64_t keys[WAVES_PER_BLOCK][HASH];
  __shared__ __align__(16) char accs_raw[WAVES_PER_BLOCK * HASH * sizeof(T)];
  auto accs = reinterpret_cast<T(*)[HASH]>(accs_raw);
  // compacted (col, val) pairs; count <= HASH/2 by the ub binning
  __shared__ int64_t ckeys_s[WAVES_PER_BLOCK][HASH / 2 + 1];
  __shared__ __align__(16) char cvals_raw[WAVES_PER_BLOCK * (HASH / 2 + 1) * sizeof(T)];
  auto ckeys = ckeys_s;
  auto cvals = reinterpret_cast<T(*)[HASH / 2 + 1]>(cvals_raw);
  __shared__ int slots[WAVES_PER_BLOCK];
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int64_t li = (int64_t)blockIdx.x * WAVES_PER_BLOCK + wid;
  if (li >= nrows) return;
  const int64_t r = rowlist[li];
  for (int i = lane; i < HASH; i += WAVE) {
    keys[wid][i] = -1;
    accs[wid][i] = ZeroOf<T>::value();
  }
  if (lane == 0) slots[wid] = 0;
  __builtin_amdgcn_wave_barrier();
  __threadfence_block();
  const int64_t as = aip[r], ae = aip[r + 1];
  for (int64_t p = as + lane; p < ae; p += WAVE) {
    const int64_t brow = (int64_t)aix[p] - a_col_lo;
    const T aval = av[p];
    const int64_t bs = bip[brow], be = bip[brow + 1];
    for (int64_t q = bs; q < be; ++q) {
      int64_t c = (int64_t)bix[q];
      uint32_t h = hash_mul(c) & (HASH - 1);
      while (true) {
        int64_t old = atomicCAS((unsigned long long*)&keys[wid][h],
                                (unsigned long long)(-1ll),
                                (unsigned long long)c);
        if (old == -1ll || old == c) {
          atomic_add_any(&accs[wid][h], aval * bv[q]);
          break;
        }
        h = (h + 1) & (HASH - 1);
      }
    }
  }
  __builtin_amdgcn_wave_barrier();
  __threadfence_block();
  // compact into [0, count) then LDS bitonic sort by column — emitting the
  // row SORTED skips the wrapper's global argsort (2 full passes over C).
  const int64_t base = cip[r];
  const int count = (int)(cip[r + 1] - base);
  for (int i = lane; i < HASH; i += WAVE) {
    if (keys[wid][i] != -1) {
      int slot = atomicAdd(&slots[wid], 1);
      ckeys[wid][slot] = keys[wid][i];
      cvals[wid][slot] = accs[wid][i];
    }
  }
  __builtin_amdgcn_wave_barrier();
  __threadfence_block();
  // pad to the next power of two with +inf sentinels
  int np2 = 1;
  while (np2 < count) np2 <<= 1;
  for (int i = count + lane; i < np2; i += WAVE) {
    ckeys[wid][i] = INT64_MAX;
  }
  __builtin_amdgcn_wave_barrier();
  __threadfence_block();
  for (int k = 2; k <= np2; k <<= 1) {
    for (int j = k >> 1; j > 0; j >>= 1) {
      for (int i = lane; i < np2; i += WAVE) {
        int ixj = i ^ j;
        if (ixj > i) {
          bool up = (i & k) == 0;
          int64_t ki = ckeys[wid][i], kj = ckeys[wid][ixj];
          if ((ki > kj) == up) {
            ckeys[wid][i] = kj;
            ckeys[wid][ixj] = ki;
            T tv = cvals[wid][i];
            cvals[wid][i] = cvals[wid][ixj];
            cvals[wid][ixj] = tv;
          }
        }
      }
      __builtin_amdgcn_wave_barrier();
      __threadfence_block();
    }
  }
  for (int i = lane; i < count; i += WAVE) {
    cix[base + i] = (index_t)ckeys[wid][i];
    cv[base + i] = cvals[wid][i];
  }
}

}  // namespace

void spgemm_nnz_hip(at::Tensor aip, at::Tensor aix, at::Tensor bip,
                    at::Tensor bix, at::Tensor rowlist, at::Tensor nnz_out,
                    int64_t a_col_lo, int64_t hash_size) {
  int64_t nrows = rowlist.numel();
  if (nrows == 0) return;
  int64_t nb = (nrows + WAVES_PER_BLOCK - 1) / WAVES_PER_BLOCK;
  DISPATCH_INDEX(aix.scalar_type(), "spgemm_nnz", [&] {
    auto launch = [&](auto kern) {
      hipLaunchKernelGGL(kern, dim3(nb), dim3(256), 0, cur_stream(),
                         aip.data_ptr<int64_t>(), aix.data_ptr<index_t>(),
                         bip.data_ptr<int64_t>(), bix.data_ptr<index_t>(),
                         rowlist.data_ptr<int64_t>(), nrows,
                         nnz_out.data_ptr<int64_t>(), a_col_lo);
    };
    if (hash_size <= 64) launch(spgemm_nnz_kernel<index_t, 64>);
    else if (hash_size <= 256) launch(spgemm_nnz_kernel<index_t, 256>);
    else launch(spgemm_nnz_kernel<index_t, 1024>);
  });
}

void spgemm_compute_hip(at::Tensor aip, at::Tensor aix, at::Tensor av,
                        at::Tensor bip, at::Tensor bix, at::Tensor bv,
                        at::Tensor rowlist, at::Tensor cip, at::Tensor cix,
                        at::Tensor cv, int64_t a_col_lo, int64_t hash_size) {
  int64_t nrows = rowlist.numel();
  if (nrows == 0) return;
  int64_t nb = (nrows + WAVES_PER_BLOCK - 1) / WAVES_PER_BLOCK;
  DISPATCH_VALUES(cv.scalar_type(), "spgemm_compute", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(aix.scalar_type(), "spgemm_compute_idx", [&] {
      auto launch = [&](auto kern) {
        hipLaunchKernelGGL(kern, dim3(nb), dim3(256), 0, cur_stream(),
                           aip.data_ptr<int64_t>(), aix.data_ptr<index_t>(),
                           av.data_ptr<T>(), bip.data_ptr<int64_t>(),
                           bix.data_ptr<index_t>(), bv.data_ptr<T>(),
                           rowlist.data_ptr<int64_t>(), nrows,
                           cip.data_ptr<int64_t>(), cix.data_ptr<index_t>(),
                           cv.data_ptr<T>(), a_col_lo);
      };
      if (hash_size <= 64) launch(spgemm_compute_kernel<T, index_t, 64>);
      else if (hash_size <= 256) launch(spgemm_compute_kernel<T, index_t, 256>);
      else launch(spgemm_compute_kernel<T, index_t, 1024>);
    });
  });
}
