// SpGEMM: two-phase Gustavson with per-row LDS hash accumulators,
// size-binned and sub-wave packed.
//
// Reference parity: SPGEMM_CSR_CSR_CSR_GPU (spgemm_csr_csr_csr.cu:33-272,
// cuSPARSE-backed there) and the CPU Gustavson semantics
// (spgemm_csr_csr_csr.cc:27-85).  MI355X design:
//  - rows BINNED by product upper bound (sum of touched B-row sizes) into
//    hash sizes 64/256/1024 so short stencil rows don't pay big table
//    inits; rows with ub > 512 use the wrapper's expand-sort-reduce;
//  - the 64-entry bin packs FOUR rows per wave (16 lanes each) — a 5-nnz
//    stencil row otherwise leaves 59 of 64 lanes idle;
//  - rows are emitted SORTED via an LDS bitonic over the compacted entries
//    (no global argsort pass over C).
#include "common.h"

namespace {

constexpr int WAVES_PER_BLOCK = 4;  // 256 threads

__device__ __forceinline__ uint32_t hash_mul(int64_t c) {
  return (uint32_t)((uint64_t)c * 2654435761u);
}

// LPR = lanes per row (16 or 64); WPB = waves per block;
// rows per block = WPB*(WAVE/LPR)
// CHECKED: rows are binned by PRODUCT upper bound, but dense-product rows
// often have far fewer distinct columns than products; the checked bin
// admits such rows speculatively and flags table overflow (nnz_out = -1)
// so the wrapper reroutes only true overflows to the ESC fallback.
template <typename index_t, int HASH, int LPR, int WPB = WAVES_PER_BLOCK,
          bool CHECKED = false>
__global__ __launch_bounds__(256) void spgemm_nnz_kernel(
    const int64_t* __restrict__ aip, const index_t* __restrict__ aix,
    const int64_t* __restrict__ bip, const index_t* __restrict__ bix,
    const int64_t* __restrict__ rowlist, int64_t nrows,
    int64_t* __restrict__ nnz_out, int64_t a_col_lo) {
  constexpr int RPB = WPB * (WAVE / LPR);
  __shared__ int64_t keys[RPB][HASH];
  __shared__ int counts[2 * RPB];  // [0,RPB): distinct-col counts; [RPB,2RPB): scan totals
  __shared__ int ovf[RPB];
  const int slot_id = threadIdx.x / LPR;  // row slot within block
  const int sl = threadIdx.x % LPR;       // lane within row
  const int64_t li = (int64_t)blockIdx.x * RPB + slot_id;
  if (li >= nrows) return;
  const int64_t r = rowlist[li];
  for (int i = sl; i < HASH; i += LPR) keys[slot_id][i] = -1;
  if (sl == 0) counts[slot_id] = 0;
  if (sl == 0) ovf[slot_id] = 0;
  __builtin_amdgcn_wave_barrier();
  __threadfence_block();
  // product-per-lane: tile the A row by LPR entries; an LDS inclusive scan
  // of the touched B-row sizes lets every lane take one (a,b) product —
  // no idle lanes when B rows are short/uneven.
  __shared__ int64_t bstart_s[RPB][LPR];
  __shared__ int bcnt_s[RPB][LPR + 1];
  const int64_t as = aip[r], ae = aip[r + 1];
  for (int64_t t0 = as; t0 < ae; t0 += LPR) {
    if (CHECKED && ovf[slot_id]) break;
    const int na = (int)min((int64_t)LPR, ae - t0);
    if (sl < na) {
      const int64_t brow = (int64_t)aix[t0 + sl] - a_col_lo;
      bstart_s[slot_id][sl] = bip[brow];
      bcnt_s[slot_id][sl] = (int)(bip[brow + 1] - bip[brow]);
    }
    __builtin_amdgcn_wave_barrier();
    __threadfence_block();
    if (sl == 0) {
      int acc = 0;
      for (int j2 = 0; j2 < na; ++j2) {
        int c0 = bcnt_s[slot_id][j2];
        bcnt_s[slot_id][j2] = acc;  // exclusive prefix
        acc += c0;
      }
      counts[slot_id + RPB] = acc;  // total, stashed past counts[]
    }
    __builtin_amdgcn_wave_barrier();
    __threadfence_block();
    const int total = counts[slot_id + RPB];
    for (int t = sl; t < total; t += LPR) {
      if (CHECKED && ovf[slot_id]) break;
      // find j: largest with prefix[j] <= t (linear ok: na <= LPR)
      int j = 0;
      for (int j2 = 1; j2 < na; ++j2) j += (bcnt_s[slot_id][j2] <= t);
      const int64_t q = bstart_s[slot_id][j] + (t - bcnt_s[slot_id][j]);
      int64_t c = (int64_t)bix[q];
      uint32_t h = hash_mul(c) & (HASH - 1);
      int probes = 0;
      while (true) {
        int64_t old = atomicCAS((unsigned long long*)&keys[slot_id][h],
                                (unsigned long long)(-1ll),
                                (unsigned long long)c);
        if (old == -1ll) { atomicAdd(&counts[slot_id], 1); break; }
        if (old == c) break;
        h = (h + 1) & (HASH - 1);
        if (CHECKED && ++probes >= HASH) { ovf[slot_id] = 1; break; }
      }
    }
    __builtin_amdgcn_wave_barrier();
    __threadfence_block();
  }
  // distinct count must leave sort headroom in the compute phase
  // (compaction arrays hold HASH/2 entries)
  if (sl == 0) {
    const bool over = CHECKED && (ovf[slot_id] || counts[slot_id] > HASH / 2);
    nnz_out[r] = over ? -1 : (int64_t)counts[slot_id];
  }
}

template <typename T, typename index_t, int HASH, int LPR,
          int WPB = WAVES_PER_BLOCK>
__global__ __launch_bounds__(256) void spgemm_compute_kernel(
    const int64_t* __restrict__ aip, const index_t* __restrict__ aix,
    const T* __restrict__ av, const int64_t* __restrict__ bip,
    const index_t* __restrict__ bix, const T* __restrict__ bv,
    const int64_t* __restrict__ rowlist, int64_t nrows,
    const int64_t* __restrict__ cip, index_t* __restrict__ cix,
    T* __restrict__ cv, int64_t a_col_lo) {
  constexpr int RPB = WPB * (WAVE / LPR);
  __shared__ int64_t keys[RPB][HASH];
  __shared__ __align__(16) char accs_raw[RPB * HASH * sizeof(T)];
  auto accs = reinterpret_cast<T(*)[HASH]>(accs_raw);
  // compacted (col, val) pairs; count <= HASH/2 by the ub binning
  __shared__ int64_t ckeys[RPB][HASH / 2 + 1];
  __shared__ __align__(16) char cvals_raw[RPB * (HASH / 2 + 1) * sizeof(T)];
  auto cvals = reinterpret_cast<T(*)[HASH / 2 + 1]>(cvals_raw);
  __shared__ int slots[RPB];
  const int slot_id = threadIdx.x / LPR;
  const int sl = threadIdx.x % LPR;
  const int64_t li = (int64_t)blockIdx.x * RPB + slot_id;
  if (li >= nrows) return;
  const int64_t r = rowlist[li];
  for (int i = sl; i < HASH; i += LPR) {
    keys[slot_id][i] = -1;
    accs[slot_id][i] = ZeroOf<T>::value();
  }
  if (sl == 0) slots[slot_id] = 0;
  __builtin_amdgcn_wave_barrier();
  __threadfence_block();
  __shared__ int64_t bstart_s[RPB][LPR];
  __shared__ int bcnt_s[RPB][LPR + 1];
  __shared__ int tot_s[RPB];
  __shared__ __align__(16) char aval_raw[RPB * LPR * sizeof(T)];
  auto aval_s = reinterpret_cast<T(*)[LPR]>(aval_raw);
  const int64_t as = aip[r], ae = aip[r + 1];
  for (int64_t t0 = as; t0 < ae; t0 += LPR) {
    const int na = (int)min((int64_t)LPR, ae - t0);
    if (sl < na) {
      const int64_t brow = (int64_t)aix[t0 + sl] - a_col_lo;
      bstart_s[slot_id][sl] = bip[brow];
      bcnt_s[slot_id][sl] = (int)(bip[brow + 1] - bip[brow]);
      aval_s[slot_id][sl] = av[t0 + sl];
    }
    __builtin_amdgcn_wave_barrier();
    __threadfence_block();
    if (sl == 0) {
      int acc = 0;
      for (int j2 = 0; j2 < na; ++j2) {
        int c0 = bcnt_s[slot_id][j2];
        bcnt_s[slot_id][j2] = acc;
        acc += c0;
      }
      tot_s[slot_id] = acc;
    }
    __builtin_amdgcn_wave_barrier();
    __threadfence_block();
    const int total = tot_s[slot_id];
    for (int t = sl; t < total; t += LPR) {
      int j = 0;
      for (int j2 = 1; j2 < na; ++j2) j += (bcnt_s[slot_id][j2] <= t);
      const int64_t q = bstart_s[slot_id][j] + (t - bcnt_s[slot_id][j]);
      int64_t c = (int64_t)bix[q];
      uint32_t h = hash_mul(c) & (HASH - 1);
      while (true) {
        int64_t old = atomicCAS((unsigned long long*)&keys[slot_id][h],
                                (unsigned long long)(-1ll),
                                (unsigned long long)c);
        if (old == -1ll || old == c) {
          atomic_add_any(&accs[slot_id][h], aval_s[slot_id][j] * bv[q]);
          break;
        }
        h = (h + 1) & (HASH - 1);
      }
    }
    __builtin_amdgcn_wave_barrier();
    __threadfence_block();
  }
  // compact into [0, count) then LDS bitonic sort by column — emitting the
  // row SORTED skips a global argsort pass over C.
  const int64_t base = cip[r];
  const int count = (int)(cip[r + 1] - base);
  for (int i = sl; i < HASH; i += LPR) {
    if (keys[slot_id][i] != -1) {
      int slot = atomicAdd(&slots[slot_id], 1);
      ckeys[slot_id][slot] = keys[slot_id][i];
      cvals[slot_id][slot] = accs[slot_id][i];
    }
  }
  __builtin_amdgcn_wave_barrier();
  __threadfence_block();
  int np2 = 1;
  while (np2 < count) np2 <<= 1;
  for (int i = count + sl; i < np2; i += LPR) {
    ckeys[slot_id][i] = INT64_MAX;
  }
  __builtin_amdgcn_wave_barrier();
  __threadfence_block();
  for (int k = 2; k <= np2; k <<= 1) {
    for (int j = k >> 1; j > 0; j >>= 1) {
      for (int i = sl; i < np2; i += LPR) {
        int ixj = i ^ j;
        if (ixj > i) {
          bool up = (i & k) == 0;
          int64_t ki = ckeys[slot_id][i], kj = ckeys[slot_id][ixj];
          if ((ki > kj) == up) {
            ckeys[slot_id][i] = kj;
            ckeys[slot_id][ixj] = ki;
            T tv = cvals[slot_id][i];
            cvals[slot_id][i] = cvals[slot_id][ixj];
            cvals[slot_id][ixj] = tv;
          }
        }
      }
      __builtin_amdgcn_wave_barrier();
      __threadfence_block();
    }
  }
  for (int i = sl; i < count; i += LPR) {
    cix[base + i] = (index_t)ckeys[slot_id][i];
    cv[base + i] = cvals[slot_id][i];
  }
}

}  // namespace

void spgemm_nnz_hip(at::Tensor aip, at::Tensor aix, at::Tensor bip,
                    at::Tensor bix, at::Tensor rowlist, at::Tensor nnz_out,
                    int64_t a_col_lo, int64_t hash_size) {
  int64_t nrows = rowlist.numel();
  if (nrows == 0) return;
  DISPATCH_INDEX(aix.scalar_type(), "spgemm_nnz", [&] {
    auto launch = [&](auto kern, int rpb, int threads) {
      hipLaunchKernelGGL(kern, dim3((nrows + rpb - 1) / rpb), dim3(threads), 0,
                         cur_stream(), aip.data_ptr<int64_t>(),
                         aix.data_ptr<index_t>(), bip.data_ptr<int64_t>(),
                         bix.data_ptr<index_t>(), rowlist.data_ptr<int64_t>(),
                         nrows, nnz_out.data_ptr<int64_t>(), a_col_lo);
    };
    if (hash_size <= 64) launch(spgemm_nnz_kernel<index_t, 64, 16>, 16, 256);
    else if (hash_size <= 256) launch(spgemm_nnz_kernel<index_t, 256, 64>, 4, 256);
    else if (hash_size <= 1024) launch(spgemm_nnz_kernel<index_t, 1024, 64>, 4, 256);
    else if (hash_size <= 2048) launch(spgemm_nnz_kernel<index_t, 2048, 64, 2>, 2, 128);
    else  // checked speculative bin: overflow -> nnz_out = -1
      launch(spgemm_nnz_kernel<index_t, 4096, 64, 1, true>, 1, 64);
  });
}

void spgemm_compute_hip(at::Tensor aip, at::Tensor aix, at::Tensor av,
                        at::Tensor bip, at::Tensor bix, at::Tensor bv,
                        at::Tensor rowlist, at::Tensor cip, at::Tensor cix,
                        at::Tensor cv, int64_t a_col_lo, int64_t hash_size) {
  int64_t nrows = rowlist.numel();
  if (nrows == 0) return;
  DISPATCH_VALUES(cv.scalar_type(), "spgemm_compute", [&] {
    using T = scalar_t;
    DISPATCH_INDEX(aix.scalar_type(), "spgemm_compute_idx", [&] {
      auto launch = [&](auto kern, int rpb, int threads) {
        hipLaunchKernelGGL(kern, dim3((nrows + rpb - 1) / rpb), dim3(threads), 0,
                           cur_stream(), aip.data_ptr<int64_t>(),
                           aix.data_ptr<index_t>(), av.data_ptr<T>(),
                           bip.data_ptr<int64_t>(), bix.data_ptr<index_t>(),
                           bv.data_ptr<T>(), rowlist.data_ptr<int64_t>(),
                           nrows, cip.data_ptr<int64_t>(),
                           cix.data_ptr<index_t>(), cv.data_ptr<T>(), a_col_lo);
      };
      if (hash_size <= 64) {
        launch(spgemm_compute_kernel<T, index_t, 64, 16>, 16, 256);
      } else if (hash_size > 2048) {
        if constexpr (sizeof(T) <= 8) {
          launch(spgemm_compute_kernel<T, index_t, 4096, 64, 1>, 1, 64);
        } else {
          TORCH_CHECK(false,
                      "spgemm: the checked 4096 bin supports 8-byte value "
                      "types only (wrapper routes complex to ESC)");
        }
      } else if (hash_size <= 256) {
        launch(spgemm_compute_kernel<T, index_t, 256, 64>, 4, 256);
      } else if (hash_size <= 1024) {
        launch(spgemm_compute_kernel<T, index_t, 1024, 64>, 4, 256);
      } else {
        launch(spgemm_compute_kernel<T, index_t, 2048, 64, 2>, 2, 128);
      }
    });
  });
}
