// DIA SpMV nt-load A/B microbenchmark (no torch; hipcc direct).
//
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/dia_nt_bench.hip -o gpurun_out/dia_nt_bench
//   ./dia_nt_bench [nx]
//
// 5-pt Poisson nx*nx in DIA layout (5 value planes, column-major), fp64.
// Variants interleaved in one process:
//   v0: plain loads (the shipping dia_spmv_kernel addressing)
//   v1: __builtin_nontemporal_load on the dvals planes (used exactly once —
//       keep them out of L2 so the reusable x lines survive)
//   v2: v1 + nontemporal store of y
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define CHECK(c)                                                      \
  do {                                                                \
    hipError_t e = (c);                                               \
    if (e != hipSuccess) {                                            \
      printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
      exit(1);                                                        \
    }                                                                 \
  } while (0)

constexpr int BLK = 256;

struct DPair { double a, b; };

__global__ void gen_dia(double* dvals, double* x, int64_t nx, int64_t N,
                        int64_t mp) {
  int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (r >= mp) return;
  // planes k=0..4: offsets -nx,-1,0,+1,+nx
  const int64_t offs[5] = {-nx, -1, 0, 1, nx};
  int64_t ix = r % nx;
  for (int k = 0; k < 5; ++k) {
    double v = 0.0;
    if (r < N) {
      int64_t c = r + offs[k];
      bool ok = c >= 0 && c < N;
      if (k == 1 && ix == 0) ok = false;
      if (k == 3 && ix == nx - 1) ok = false;
      if (ok) v = (k == 2) ? 4.0 : -1.0;
    }
    dvals[(int64_t)k * mp + r] = v;
  }
  if (r < N) x[r] = 1.0 + (double)(r % 97) * 0.01;
}

__device__ __forceinline__ int64_t swz(int64_t bid, int64_t nwg) {
  int64_t q = nwg / 8, rr = nwg % 8;
  int64_t xcd = bid % 8, idx = bid / 8;
  return (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
}

template <int NT>
__global__ __launch_bounds__(BLK) void dia_v(const double* __restrict__ dvals,
                                             const int64_t* __restrict__ offs,
                                             const double* __restrict__ x,
                                             double* __restrict__ y,
                                             int64_t m, int64_t mp, int W) {
  const int64_t b = NT >= 3 ? swz(blockIdx.x, gridDim.x) : blockIdx.x;
  const int64_t t = b * BLK + threadIdx.x;
  const int64_t r0 = 2 * t;
  if (r0 >= mp) return;
  double a0 = 0.0, a1 = 0.0;
  for (int k = 0; k < W; ++k) {
    const int64_t base = (int64_t)k * mp + r0;
    DPair vv;
    if (NT >= 1) {
      vv.a = __builtin_nontemporal_load(&dvals[base]);
      vv.b = __builtin_nontemporal_load(&dvals[base + 1]);
    } else {
      vv = *reinterpret_cast<const DPair*>(&dvals[base]);
    }
    const int64_t c0 = r0 + offs[k];
    const int64_t i0 = min(max(c0, (int64_t)0), m - 1);
    const int64_t i1 = min(max(c0 + 1, (int64_t)0), m - 1);
    a0 += vv.a * x[i0];
    a1 += vv.b * x[i1];
  }
  if (r0 + 1 < m) {
    if (NT == 2) {
      __builtin_nontemporal_store(a0, &y[r0]);
      __builtin_nontemporal_store(a1, &y[r0 + 1]);
    } else {
      DPair out{a0, a1};
      *reinterpret_cast<DPair*>(&y[r0]) = out;
    }
  } else if (r0 < m) {
    y[r0] = a0;
  }
}

int main(int argc, char** argv) {
  int64_t nx = argc > 1 ? atoll(argv[1]) : 16384;
  int64_t N = nx * nx;
  int64_t mp = (N + 1) & ~1LL;
  const int W = 5;
  double *dvals, *x, *y;
  int64_t* offs;
  CHECK(hipMalloc(&dvals, (size_t)W * mp * 8));
  CHECK(hipMalloc(&x, (size_t)N * 8));
  CHECK(hipMalloc(&y, (size_t)N * 8));
  CHECK(hipMalloc(&offs, W * 8));
  int64_t h_offs[5] = {-nx, -1, 0, 1, nx};
  CHECK(hipMemcpy(offs, h_offs, W * 8, hipMemcpyHostToDevice));
  gen_dia<<<(mp + 255) / 256, 256>>>(dvals, x, nx, N, mp);
  CHECK(hipDeviceSynchronize());

  const int64_t nblocks = (mp / 2 + BLK - 1) / BLK;
  const int iters = 30;
  double bytes = (double)W * mp * 8 + 2.0 * N * 8;
  const char* names[4] = {"v0_plain", "v1_nt_vals", "v2_nt_vals_y",
                          "v3_nt_swz"};
  std::vector<std::vector<float>> ms(4);
  hipEvent_t e0, e1;
  CHECK(hipEventCreate(&e0));
  CHECK(hipEventCreate(&e1));
  for (int rep = 0; rep < iters; ++rep) {
    for (int v = 0; v < 4; ++v) {
      CHECK(hipEventRecord(e0));
      switch (v) {
        case 0: dia_v<0><<<nblocks, BLK>>>(dvals, offs, x, y, N, mp, W); break;
        case 1: dia_v<1><<<nblocks, BLK>>>(dvals, offs, x, y, N, mp, W); break;
        case 2: dia_v<2><<<nblocks, BLK>>>(dvals, offs, x, y, N, mp, W); break;
        case 3: dia_v<3><<<nblocks, BLK>>>(dvals, offs, x, y, N, mp, W); break;
      }
      CHECK(hipEventRecord(e1));
      CHECK(hipEventSynchronize(e1));
      float t;
      CHECK(hipEventElapsedTime(&t, e0, e1));
      if (rep >= 5) ms[v].push_back(t);
    }
  }
  // checksum parity
  for (int v = 0; v < 4; ++v) {
    std::sort(ms[v].begin(), ms[v].end());
    float med = ms[v][ms[v].size() / 2];
    printf("%-14s median %.3f ms  %.0f GB/s\n", names[v], med,
           bytes / (med * 1e-3) / 1e9);
  }
  std::vector<double> hy(16);
  CHECK(hipMemcpy(hy.data(), y, 16 * 8, hipMemcpyDeviceToHost));
  printf("y[0..3] = %.3f %.3f %.3f %.3f\n", hy[0], hy[1], hy[2], hy[3]);
  return 0;
}
