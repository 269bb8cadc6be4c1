// Standalone SpMV variant A/B microbenchmark (no torch; hipcc direct).
//
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/spmv_bench.hip -o gpurun_out/spmv_bench
//   ./spmv_bench [nx]
//
// Generates the 5-pt Poisson CSR (fp64, int32 idx) on-device and times
// kernel variants interleaved in one process (guide §5.4 rule 24), printing
// median ms and effective GB/s per variant.
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define CHECK(c)                                                     \
  do {                                                               \
    hipError_t e = (c);                                              \
    if (e != hipSuccess) {                                           \
      printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
      exit(1);                                                       \
    }                                                                \
  } while (0)

constexpr int BLK = 256;

__device__ __forceinline__ int64_t lbi(const int64_t* a, int64_t n, int64_t k) {
  int64_t lo = 0, hi = n;
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    if (a[mid] < k) lo = mid + 1; else hi = mid;
  }
  return lo;
}

__device__ __forceinline__ int64_t swz(int64_t bid, int64_t nwg) {
  int64_t q = nwg / 8, rr = nwg % 8;
  int64_t xcd = bid % 8, idx = bid / 8;
  return (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
}

// ---- matrix generation: 5-pt Poisson nx*nx --------------------------------
__global__ void gen_kernel(int64_t* indptr, int* indices, double* vals,
                           int64_t nx, int64_t N) {
  int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (r > N) return;
  int64_t ix = r % nx;
  // count pattern for row r
  auto cnt = [&](int64_t rr) {
    int64_t c = 1;
    int64_t jx = rr % nx;
    if (rr - nx >= 0) ++c;
    if (jx > 0) ++c;
    if (jx < nx - 1) ++c;
    if (rr + nx < N) ++c;
    return c;
  };
  // indptr via closed form: total = 5N - 2*nx(border rows missing vert) - 2*(nx rows missing horz)... compute directly
  // easier: indptr[r] = 5r - (#missing before r). missing west: rows with ix==0 -> r/nx terms... do arithmetic:
  int64_t k = r;
  int64_t full = 5 * k;
  int64_t miss_n = min(k, nx);                 // rows 0..nx-1 have no north
  int64_t miss_s = max((int64_t)0, k - (N - nx));  // rows N-nx..N-1 no south
  int64_t rows_w = (k + nx - 1) / nx;          // rows with ix==0 among 0..k-1
  int64_t rows_e = k / nx;                     // rows with ix==nx-1 among 0..k-1
  indptr[r] = full - miss_n - miss_s - rows_w - rows_e;
  if (r == N) return;
  int64_t o = indptr[r];
  if (r - nx >= 0) { indices[o] = (int)(r - nx); vals[o] = -1.0; ++o; }
  if (ix > 0) { indices[o] = (int)(r - 1); vals[o] = -1.0; ++o; }
  indices[o] = (int)r; vals[o] = 4.0; ++o;
  if (ix < nx - 1) { indices[o] = (int)(r + 1); vals[o] = -1.0; ++o; }
  if (r + nx < N) { indices[o] = (int)(r + nx); vals[o] = -1.0; ++o; }
}

// ---- V0: pure stream ceiling (vals+idx only) ------------------------------
__global__ __launch_bounds__(BLK) void v0_stream(const int* __restrict__ idx,
                                                 const double* __restrict__ vals,
                                                 double* __restrict__ sink,
                                                 int64_t nnz) {
  int64_t i0 = ((int64_t)blockIdx.x * BLK + threadIdx.x) * 4;
  double acc = 0;
  int64_t stride = (int64_t)gridDim.x * BLK * 4;
  for (int64_t i = i0; i + 3 < nnz; i += stride) {
    const int4 ii = *reinterpret_cast<const int4*>(&idx[i]);
    const double2 a = *reinterpret_cast<const double2*>(&vals[i]);
    const double2 b = *reinterpret_cast<const double2*>(&vals[i + 2]);
    acc += a.x + a.y + b.x + b.y + ii.x + ii.y + ii.z + ii.w;
  }
  if (acc == 12345.678) sink[0] = acc;
}

// ---- V0g: stream + x gather ------------------------------------------------
__global__ __launch_bounds__(BLK) void v0_gather(const int* __restrict__ idx,
                                                 const double* __restrict__ vals,
                                                 const double* __restrict__ x,
                                                 double* __restrict__ sink,
                                                 int64_t nnz) {
  int64_t i0 = ((int64_t)blockIdx.x * BLK + threadIdx.x) * 4;
  double acc = 0;
  int64_t stride = (int64_t)gridDim.x * BLK * 4;
  for (int64_t i = i0; i + 3 < nnz; i += stride) {
    const int4 ii = *reinterpret_cast<const int4*>(&idx[i]);
    const double2 a = *reinterpret_cast<const double2*>(&vals[i]);
    const double2 b = *reinterpret_cast<const double2*>(&vals[i + 2]);
    acc += a.x * x[ii.x] + a.y * x[ii.y] + b.x * x[ii.z] + b.y * x[ii.w];
  }
  if (acc == 12345.678) sink[0] = acc;
}

// ---- V1: thread per row ----------------------------------------------------
template <bool SWZ>
__global__ __launch_bounds__(BLK) void v1_row(const int64_t* __restrict__ indptr,
                                              const int* __restrict__ idx,
                                              const double* __restrict__ vals,
                                              const double* __restrict__ x,
                                              double* __restrict__ y, int64_t m) {
  int64_t b = SWZ ? swz(blockIdx.x, gridDim.x) : blockIdx.x;
  int64_t r = b * BLK + threadIdx.x;
  if (r >= m) return;
  int64_t e = indptr[r + 1];
  double acc = 0;
  for (int64_t p = indptr[r]; p < e; ++p) acc += vals[p] * x[idx[p]];
  y[r] = acc;
}

// ---- V2/V3: nnz-split LDS-staged, scalar vs quad loads ---------------------
template <int VT, bool QUAD, bool SWZ>
__global__ __launch_bounds__(BLK) void v23_nnz(const int64_t* __restrict__ indptr,
                                               const int* __restrict__ idx,
                                               const double* __restrict__ vals,
                                               const double* __restrict__ x,
                                               double* __restrict__ y,
                                               double* __restrict__ carry_val,
                                               int64_t* __restrict__ carry_row,
                                               int64_t m, int64_t nnz) {
  constexpr int64_t NPB = (int64_t)BLK * VT;
  extern __shared__ double prod[];
  __shared__ double red[BLK];
  const int64_t b = SWZ ? swz(blockIdx.x, gridDim.x) : blockIdx.x;
  const int64_t s = b * NPB;
  const int64_t e = min(s + NPB, nnz);
  const int tid = threadIdx.x;
  const int64_t ro0 = lbi(indptr, m, s);
  const int64_t ro1 = (e == nnz) ? m : lbi(indptr, m, e);
  if (e - s == NPB) {
    if (QUAD) {
      int4 i4[VT / 4];
      double2 v2[VT / 2];
#pragma unroll
      for (int k = 0; k < VT / 4; ++k)
        i4[k] = *reinterpret_cast<const int4*>(&idx[s + tid * 4 + (int64_t)k * 4 * BLK]);
#pragma unroll
      for (int k = 0; k < VT / 4; ++k) {
        v2[2 * k] = *reinterpret_cast<const double2*>(&vals[s + tid * 4 + (int64_t)k * 4 * BLK]);
        v2[2 * k + 1] = *reinterpret_cast<const double2*>(&vals[s + tid * 4 + (int64_t)k * 4 * BLK + 2]);
      }
      double xv[VT];
#pragma unroll
      for (int k = 0; k < VT / 4; ++k) {
        xv[4 * k + 0] = x[i4[k].x];
        xv[4 * k + 1] = x[i4[k].y];
        xv[4 * k + 2] = x[i4[k].z];
        xv[4 * k + 3] = x[i4[k].w];
      }
#pragma unroll
      for (int k = 0; k < VT / 4; ++k) {
        double2 p0{v2[2 * k].x * xv[4 * k], v2[2 * k].y * xv[4 * k + 1]};
        double2 p1{v2[2 * k + 1].x * xv[4 * k + 2], v2[2 * k + 1].y * xv[4 * k + 3]};
        *reinterpret_cast<double2*>(&prod[tid * 4 + k * 4 * BLK]) = p0;
        *reinterpret_cast<double2*>(&prod[tid * 4 + k * 4 * BLK + 2]) = p1;
      }
    } else {
      int ii[VT];
      double vv[VT];
#pragma unroll
      for (int k = 0; k < VT; ++k) ii[k] = idx[s + tid + k * BLK];
#pragma unroll
      for (int k = 0; k < VT; ++k) vv[k] = vals[s + tid + k * BLK];
      double xv[VT];
#pragma unroll
      for (int k = 0; k < VT; ++k) xv[k] = x[ii[k]];
#pragma unroll
      for (int k = 0; k < VT; ++k) prod[tid + k * BLK] = vv[k] * xv[k];
    }
  } else {
    for (int64_t i = s + tid; i < e; i += BLK) prod[i - s] = vals[i] * x[idx[i]];
  }
  __syncthreads();
  for (int64_t r = ro0 + tid; r < ro1; r += BLK) {
    int64_t rs = indptr[r];
    int64_t re = min(indptr[r + 1], e);
    double acc = 0;
    for (int64_t p = rs; p < re; ++p) acc += prod[p - s];
    y[r] = acc;
  }
  bool has_carry = false;
  if (ro0 > 0) {
    int64_t cend = (ro0 < m) ? min(indptr[ro0], e) : e;
    if (cend > s) {
      has_carry = true;
      double acc = 0;
      for (int64_t p = s + tid; p < cend; p += BLK) acc += prod[p - s];
      red[tid] = acc;
      __syncthreads();
      for (int w = BLK / 2; w > 0; w >>= 1) {
        if (tid < w) red[tid] += red[tid + w];
        __syncthreads();
      }
      if (tid == 0) { carry_val[b] = red[0]; carry_row[b] = ro0 - 1; }
    }
  }
  if (!has_carry && tid == 0) carry_row[b] = -1;
}

// ---- V5t: ELL with compile-time W (fully unrolled k-loop) -----------------
template <int WCT>
__global__ __launch_bounds__(BLK) void v5_ell_ct(const int* __restrict__ eidx,
                                                 const double* __restrict__ evals,
                                                 const double* __restrict__ x,
                                                 double* __restrict__ y,
                                                 int64_t mp) {
  int64_t t = (int64_t)blockIdx.x * BLK + threadIdx.x;
  int64_t r0 = 2 * t;
  if (r0 >= mp) return;
  double a0 = 0, a1 = 0;
#pragma unroll
  for (int k = 0; k < WCT; ++k) {
    const double2 v = *reinterpret_cast<const double2*>(&evals[(int64_t)k * mp + r0]);
    const int2 ii = *reinterpret_cast<const int2*>(&eidx[(int64_t)k * mp + r0]);
    a0 += v.x * x[ii.x];
    a1 += v.y * x[ii.y];
  }
  *reinterpret_cast<double2*>(&y[r0]) = double2{a0, a1};
}

// ---- V7: ELL 4 rows/thread ------------------------------------------------
__global__ __launch_bounds__(BLK) void v7_ell4(const int* __restrict__ eidx,
                                               const double* __restrict__ evals,
                                               const double* __restrict__ x,
                                               double* __restrict__ y,
                                               int64_t mp, int W) {
  int64_t t = (int64_t)blockIdx.x * BLK + threadIdx.x;
  int64_t r0 = 4 * t;
  if (r0 >= mp) return;
  double a0 = 0, a1 = 0, a2 = 0, a3 = 0;
  for (int k = 0; k < W; ++k) {
    const int64_t base = (int64_t)k * mp + r0;
    const double2 va = *reinterpret_cast<const double2*>(&evals[base]);
    const double2 vb = *reinterpret_cast<const double2*>(&evals[base + 2]);
    const int4 ii = *reinterpret_cast<const int4*>(&eidx[base]);
    a0 += va.x * x[ii.x];
    a1 += va.y * x[ii.y];
    a2 += vb.x * x[ii.z];
    a3 += vb.y * x[ii.w];
  }
  *reinterpret_cast<double2*>(&y[r0]) = double2{a0, a1};
  *reinterpret_cast<double2*>(&y[r0 + 2]) = double2{a2, a3};
}

// ---- V5: column-major padded ELL, 2 rows/thread, 16B loads ---------------
template <bool FUSE_DOT>
__global__ __launch_bounds__(BLK) void v5_ell(const int* __restrict__ eidx,
                                              const double* __restrict__ evals,
                                              const double* __restrict__ x,
                                              double* __restrict__ y,
                                              const double* __restrict__ p,
                                              double* __restrict__ dotp,
                                              int64_t mp, int W) {
  // mp = padded row count (even); element (k, r) at [k*mp + r]
  int64_t t = (int64_t)blockIdx.x * BLK + threadIdx.x;
  int64_t r0 = 2 * t;
  if (r0 >= mp) return;
  double a0 = 0, a1 = 0;
  for (int k = 0; k < W; ++k) {
    const double2 v = *reinterpret_cast<const double2*>(&evals[(int64_t)k * mp + r0]);
    const int2 ii = *reinterpret_cast<const int2*>(&eidx[(int64_t)k * mp + r0]);
    a0 += v.x * x[ii.x];
    a1 += v.y * x[ii.y];
  }
  *reinterpret_cast<double2*>(&y[r0]) = double2{a0, a1};
}

__global__ void fixup(const double* cv, const int64_t* cr, double* y, int64_t nb) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= nb) return;
  int64_t r = cr[i];
  if (r >= 0) atomicAdd(&y[r], cv[i]);
}

// ---- V5: wave-per-row-group "stream rows": each wave covers 64 consecutive
// rows; lanes cooperatively load the row block's nnz range coalesced and use
// lane-exchange-free direct summation via LDS per-wave. (simplified: LDS)
// (kept out of v1 for comparison of barrier-free structures)

int main(int argc, char** argv) {
  int64_t nx = argc > 1 ? atoll(argv[1]) : 8192;
  int64_t N = nx * nx;
  int64_t nnz_max = 5 * N;
  int64_t *indptr, *carry_row;
  int* indices;
  double *vals, *x, *y, *carry_val, *sink;
  CHECK(hipMalloc(&indptr, (N + 1) * sizeof(int64_t)));
  CHECK(hipMalloc(&indices, nnz_max * sizeof(int)));
  CHECK(hipMalloc(&vals, nnz_max * sizeof(double)));
  CHECK(hipMalloc(&x, N * sizeof(double)));
  CHECK(hipMalloc(&y, N * sizeof(double)));
  CHECK(hipMalloc(&sink, sizeof(double)));
  hipLaunchKernelGGL(gen_kernel, dim3((N + 256) / 256 + 1), dim3(256), 0, 0,
                     indptr, indices, vals, nx, N);
  CHECK(hipDeviceSynchronize());
  int64_t nnz;
  CHECK(hipMemcpy(&nnz, &indptr[N], sizeof(int64_t), hipMemcpyDeviceToHost));
  printf("nx=%ld N=%ld nnz=%ld\n", (long)nx, (long)N, (long)nnz);
  // x = 1..  (init via kernel)
  CHECK(hipMemset(y, 0, N * sizeof(double)));
  {
    std::vector<double> hx(N, 1.0);
    for (int64_t i = 0; i < N; i += 7) hx[i] = 1.5;
    CHECK(hipMemcpy(x, hx.data(), N * sizeof(double), hipMemcpyHostToDevice));
  }
  int64_t nb8 = (nnz + 2047) / 2048, nb16 = (nnz + 4095) / 4096;
  // build ELL mirror on host (W=5)
  int W = 5;
  int64_t mp = (N + 3) & ~3ll;
  int* eidx; double* evals;
  CHECK(hipMalloc(&eidx, W * mp * sizeof(int)));
  CHECK(hipMalloc(&evals, W * mp * sizeof(double)));
  {
    std::vector<int64_t> hip_(N + 1);
    std::vector<int> hix(nnz), hei((size_t)W * mp, 0);
    std::vector<double> hv(nnz), hev((size_t)W * mp, 0.0);
    CHECK(hipMemcpy(hip_.data(), indptr, (N + 1) * 8, hipMemcpyDeviceToHost));
    CHECK(hipMemcpy(hix.data(), indices, nnz * 4, hipMemcpyDeviceToHost));
    CHECK(hipMemcpy(hv.data(), vals, nnz * 8, hipMemcpyDeviceToHost));
    for (int64_t r = 0; r < N; ++r) {
      int64_t c = hip_[r + 1] - hip_[r];
      for (int64_t k = 0; k < W; ++k) {
        if (k < c) {
          hei[(size_t)k * mp + r] = hix[hip_[r] + k];
          hev[(size_t)k * mp + r] = hv[hip_[r] + k];
        } else {
          hei[(size_t)k * mp + r] = hix[hip_[r]];  // pad: first col, val 0
        }
      }
    }
    CHECK(hipMemcpy(eidx, hei.data(), (size_t)W * mp * 4, hipMemcpyHostToDevice));
    CHECK(hipMemcpy(evals, hev.data(), (size_t)W * mp * 8, hipMemcpyHostToDevice));
  }
  CHECK(hipMalloc(&carry_val, (nb8 + 1) * sizeof(double)));
  CHECK(hipMalloc(&carry_row, (nb8 + 1) * sizeof(int64_t)));

  double ref_bytes = nnz * 12.0 + N * 8.0 * 3.0;  // vals+idx+x+y+indptr(≈)

  struct V { const char* name; int id; };
  std::vector<V> vs = {{"v0_stream", 0}, {"v0_gather", 1}, {"v1_row", 2},
                       {"v1_row_swz", 3}, {"v2_scalar_swz", 4},
                       {"v3_quad_swz", 5}, {"v3_quad_noswz", 6},
                       {"v4_quad_vt16_swz", 7}, {"v5_ell", 8},
                       {"v6_ell_ctW", 9}, {"v7_ell4", 10}};
  const int ROUNDS = 7, REPS = 3;
  std::vector<std::vector<float>> times(vs.size());
  hipEvent_t t0, t1;
  hipEventCreate(&t0);
  hipEventCreate(&t1);
  int64_t sgrid = 4096;
  for (int round = 0; round < ROUNDS; ++round) {
    for (size_t vi = 0; vi < vs.size(); ++vi) {
      hipEventRecord(t0);
      for (int rep = 0; rep < REPS; ++rep) {
        switch (vs[vi].id) {
          case 0:
            hipLaunchKernelGGL(v0_stream, dim3(sgrid), dim3(BLK), 0, 0, indices, vals, sink, nnz);
            break;
          case 1:
            hipLaunchKernelGGL(v0_gather, dim3(sgrid), dim3(BLK), 0, 0, indices, vals, x, sink, nnz);
            break;
          case 2:
            hipLaunchKernelGGL((v1_row<false>), dim3((N + BLK - 1) / BLK), dim3(BLK), 0, 0, indptr, indices, vals, x, y, N);
            break;
          case 3:
            hipLaunchKernelGGL((v1_row<true>), dim3((N + BLK - 1) / BLK), dim3(BLK), 0, 0, indptr, indices, vals, x, y, N);
            break;
          case 4:
            hipLaunchKernelGGL((v23_nnz<8, false, true>), dim3(nb8), dim3(BLK), 2048 * 8, 0, indptr, indices, vals, x, y, carry_val, carry_row, N, nnz);
            hipLaunchKernelGGL(fixup, dim3((nb8 + 255) / 256), dim3(256), 0, 0, carry_val, carry_row, y, nb8);
            break;
          case 5:
            hipLaunchKernelGGL((v23_nnz<8, true, true>), dim3(nb8), dim3(BLK), 2048 * 8, 0, indptr, indices, vals, x, y, carry_val, carry_row, N, nnz);
            hipLaunchKernelGGL(fixup, dim3((nb8 + 255) / 256), dim3(256), 0, 0, carry_val, carry_row, y, nb8);
            break;
          case 6:
            hipLaunchKernelGGL((v23_nnz<8, true, false>), dim3(nb8), dim3(BLK), 2048 * 8, 0, indptr, indices, vals, x, y, carry_val, carry_row, N, nnz);
            hipLaunchKernelGGL(fixup, dim3((nb8 + 255) / 256), dim3(256), 0, 0, carry_val, carry_row, y, nb8);
            break;
          case 7:
            hipLaunchKernelGGL((v23_nnz<16, true, true>), dim3(nb16), dim3(BLK), 4096 * 8, 0, indptr, indices, vals, x, y, carry_val, carry_row, N, nnz);
            hipLaunchKernelGGL(fixup, dim3((nb16 + 255) / 256), dim3(256), 0, 0, carry_val, carry_row, y, nb16);
            break;
          case 8:
            hipLaunchKernelGGL((v5_ell<false>), dim3((mp / 2 + BLK - 1) / BLK), dim3(BLK), 0, 0, eidx, evals, x, y, nullptr, nullptr, mp, W);
            break;
          case 9:
            hipLaunchKernelGGL((v5_ell_ct<5>), dim3((mp / 2 + BLK - 1) / BLK), dim3(BLK), 0, 0, eidx, evals, x, y, mp);
            break;
          case 10:
            hipLaunchKernelGGL(v7_ell4, dim3((mp / 4 + BLK - 1) / BLK), dim3(BLK), 0, 0, eidx, evals, x, y, mp, W);
            break;
        }
      }
      hipEventRecord(t1);
      CHECK(hipEventSynchronize(t1));
      float ms;
      hipEventElapsedTime(&ms, t0, t1);
      times[vi].push_back(ms / REPS);
    }
  }
  // correctness spot-check vs v1 for v3
  {
    std::vector<double> y1(N), y3(N);
    hipLaunchKernelGGL((v1_row<false>), dim3((N + BLK - 1) / BLK), dim3(BLK), 0, 0, indptr, indices, vals, x, y, N);
    CHECK(hipMemcpy(y1.data(), y, N * sizeof(double), hipMemcpyDeviceToHost));
    hipLaunchKernelGGL((v23_nnz<8, true, true>), dim3(nb8), dim3(BLK), 2048 * 8, 0, indptr, indices, vals, x, y, carry_val, carry_row, N, nnz);
    hipLaunchKernelGGL(fixup, dim3((nb8 + 255) / 256), dim3(256), 0, 0, carry_val, carry_row, y, nb8);
    CHECK(hipMemcpy(y3.data(), y, N * sizeof(double), hipMemcpyDeviceToHost));
    double mx = 0;
    for (int64_t i = 0; i < N; ++i) mx = std::max(mx, std::abs(y1[i] - y3[i]));
    printf("max |v1 - v3| = %.3e\n", mx);
    hipLaunchKernelGGL((v5_ell<false>), dim3((mp / 2 + BLK - 1) / BLK), dim3(BLK), 0, 0, eidx, evals, x, y, nullptr, nullptr, mp, W);
    CHECK(hipMemcpy(y3.data(), y, N * sizeof(double), hipMemcpyDeviceToHost));
    mx = 0;
    for (int64_t i = 0; i < N; ++i) mx = std::max(mx, std::abs(y1[i] - y3[i]));
    printf("max |v1 - v5ell| = %.3e\n", mx);
  }
  for (size_t vi = 0; vi < vs.size(); ++vi) {
    std::sort(times[vi].begin(), times[vi].end());
    float med = times[vi][times[vi].size() / 2];
    float mn = times[vi][0];
    double bytes = (vs[vi].id <= 1) ? (nnz * 12.0) : ref_bytes;
    printf("%-18s median %8.3f ms  min %8.3f ms  eff %7.1f GB/s\n",
           vs[vi].name, med, mn, bytes / (med * 1e6));
  }
  return 0;
}
