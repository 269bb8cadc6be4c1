// MFMA SpMM data point: convert the analytic "no MFMA for
// sparse fp64" argument into a MEASURED one.
//
// Compares, at k=32/64 dense columns:
//   1. lane-tiled CSR SpMM (the production kernel's shape, spmm.hip)
//   2. 16x16-block BSR SpMM on v_mfma_f64_16x16x4f64 matrix cores
// on two structures:
//   a. 5-diagonal Poisson-like band (the flagship workload's structure)
//      -> BSR blocks are ~18%/6% filled: MFMA does ~5-14x the FLOPs AND
//         moves ~5x the bytes (dense blocks), so bandwidth-bound loses
//   b. fully dense 16x16 blocks (3 block diagonals)
//      -> the structure MFMA is built for; index overhead of CSR vanishes
// plus a register-resident MFMA peak probe (no memory traffic).
//
// Self-checks every kernel against a host reference before timing.
//
// Build: hipcc --offload-arch=gfx950 -O3 tools/mfma_spmm_bench.hip -o /tmp/mfma_bench
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>
#include <string>
#include <algorithm>

#define HIP_CHECK(x)                                                       \
  do {                                                                     \
    hipError_t e = (x);                                                    \
    if (e != hipSuccess) {                                                 \
      printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__);      \
      exit(1);                                                             \
    }                                                                      \
  } while (0)

typedef double d4 __attribute__((ext_vector_type(4)));

// ---------------------------------------------------------------- CSR SpMM
// C[r, j] = sum_p vals[p] * B[col[p], j]; 64 lanes over j, 4 rows per block
__global__ __launch_bounds__(256) void csr_spmm(
    const long* __restrict__ indptr, const int* __restrict__ indices,
    const double* __restrict__ vals, const double* __restrict__ B,
    double* __restrict__ C, long m, long k) {
  long r = (long)blockIdx.y * blockDim.y + threadIdx.y;
  long j = (long)blockIdx.x * 64 + threadIdx.x;
  if (r >= m) return;
  double acc = 0.0;
  long e = indptr[r + 1];
  for (long p = indptr[r]; p < e; ++p) {
    long c = indices[p];
    if (j < k) acc += vals[p] * B[c * k + j];
  }
  if (j < k) C[r * k + j] = acc;
}

// ---------------------------------------------------------------- BSR MFMA
// one wave computes the C tile [brow*16.. , jt*16..) via f64 16x16x4 MFMAs.
// v_mfma_f64_16x16x4f64 lane maps (measured with --probe on gfx950):
//   A[i][kk]: i = lane%16, kk = lane/16          (1 dbl / lane)
//   B[kk][j]: j = lane%16, kk = lane/16
//   D[i][j]:  j = lane%16, i = (lane/16) + 4*reg (d4 / lane)
__global__ __launch_bounds__(256) void bsr_mfma_spmm(
    const long* __restrict__ bptr, const int* __restrict__ bcol,
    const double* __restrict__ bvals, const double* __restrict__ B,
    double* __restrict__ C, long nbrows, long k) {
  const int lane = threadIdx.x & 63;
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const long tiles_j = k >> 4;
  const long brow = wave / tiles_j;
  const long jt = wave % tiles_j;
  if (brow >= nbrows) return;
  const int li = lane & 15;   // the "16" index
  const int lk = lane >> 4;   // the "4" index
  d4 acc = {0.0, 0.0, 0.0, 0.0};
  const long e = bptr[brow + 1];
  for (long blk = bptr[brow]; blk < e; ++blk) {
    const double* Ab = bvals + blk * 256;  // row-major 16x16 block
    const long c16 = (long)bcol[blk] * 16;
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      double a = Ab[li * 16 + kk * 4 + lk];                 // A[i][kk*4+lk]
      double b = B[(c16 + kk * 4 + lk) * k + jt * 16 + li]; // B[kk*4+lk][j]
      acc = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc, 0, 0, 0);
    }
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    long i = brow * 16 + lk + 4 * r;
    C[i * k + jt * 16 + li] = acc[r];
  }
}

// ------------------------------------------------------- mapping probe
// one wave, one mfma: a/b loaded under a candidate lane map; host tries
// both D interpretations and reports the combination that matches.
__global__ void map_probe(const double* A16x4, const double* B4x16,
                          double* out /*64*4*/, int amap, int bmap) {
  int l = threadIdx.x;
  int i = amap ? (l >> 4) : (l & 15);
  int ka = amap ? (l & 15) : (l >> 4);
  double a = (ka < 4) ? A16x4[i * 4 + ka] : 0.0;
  int j = bmap ? (l >> 4) : (l & 15);
  int kb = bmap ? (l & 15) : (l >> 4);
  double b = (kb < 4) ? B4x16[kb * 16 + j] : 0.0;
  d4 acc = {0.0, 0.0, 0.0, 0.0};
  acc = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc, 0, 0, 0);
  for (int r = 0; r < 4; ++r) out[l * 4 + r] = acc[r];
}

typedef float f4v __attribute__((ext_vector_type(4)));
__global__ void map_probe_f32(const float* A16x4, const float* B4x16,
                              float* out /*64*4*/, int amap, int bmap) {
  int l = threadIdx.x;
  int i = amap ? (l >> 4) : (l & 15);
  int ka = amap ? (l & 15) : (l >> 4);
  float a = (ka < 4) ? A16x4[i * 4 + ka] : 0.0f;
  int j = bmap ? (l >> 4) : (l & 15);
  int kb = bmap ? (l & 15) : (l >> 4);
  float b = (kb < 4) ? B4x16[kb * 16 + j] : 0.0f;
  f4v acc = {0.0f, 0.0f, 0.0f, 0.0f};
  acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
  for (int r = 0; r < 4; ++r) out[l * 4 + r] = acc[r];
}

void probe_mapping_f32() {
  std::vector<float> A(64), B(64);
  for (int i = 0; i < 64; ++i) {
    A[i] = 0.1f * i + 0.3f;
    B[i] = 0.05f * i - 1.1f;
  }
  float *dA, *dB, *dO;
  HIP_CHECK(hipMalloc(&dA, 64 * 4));
  HIP_CHECK(hipMalloc(&dB, 64 * 4));
  HIP_CHECK(hipMalloc(&dO, 256 * 4));
  HIP_CHECK(hipMemcpy(dA, A.data(), 64 * 4, hipMemcpyDefault));
  HIP_CHECK(hipMemcpy(dB, B.data(), 64 * 4, hipMemcpyDefault));
  float ref[16][16];
  for (int i = 0; i < 16; ++i)
    for (int j = 0; j < 16; ++j) {
      ref[i][j] = 0.0f;
      for (int k = 0; k < 4; ++k) ref[i][j] += A[i * 4 + k] * B[k * 16 + j];
    }
  for (int amap = 0; amap < 2; ++amap)
    for (int bmap = 0; bmap < 2; ++bmap) {
      hipLaunchKernelGGL(map_probe_f32, dim3(1), dim3(64), 0, 0, dA, dB, dO,
                         amap, bmap);
      HIP_CHECK(hipDeviceSynchronize());
      std::vector<float> O(256);
      HIP_CHECK(hipMemcpy(O.data(), dO, 256 * 4, hipMemcpyDefault));
      for (int dmap = 0; dmap < 2; ++dmap) {
        double err = 0.0;
        for (int l = 0; l < 64; ++l)
          for (int r = 0; r < 4; ++r) {
            int j = l & 15;
            int i = dmap ? ((l >> 4) + 4 * r) : (4 * (l >> 4) + r);
            err = std::max(err, (double)std::fabs(O[l * 4 + r] - ref[i][j]));
          }
        printf("f32 probe amap=%d bmap=%d dmap=%d err=%.3e%s\n", amap, bmap,
               dmap, err, err < 1e-4 ? "  <-- MATCH" : "");
      }
    }
  HIP_CHECK(hipFree(dA));
  HIP_CHECK(hipFree(dB));
  HIP_CHECK(hipFree(dO));
}

void probe_mapping() {
  std::vector<double> A(64), B(64);
  for (int i = 0; i < 64; ++i) {
    A[i] = 0.1 * i + 0.3;
    B[i] = 0.05 * i - 1.1;
  }
  double *dA, *dB, *dO;
  HIP_CHECK(hipMalloc(&dA, 64 * 8));
  HIP_CHECK(hipMalloc(&dB, 64 * 8));
  HIP_CHECK(hipMalloc(&dO, 256 * 8));
  HIP_CHECK(hipMemcpy(dA, A.data(), 64 * 8, hipMemcpyDefault));
  HIP_CHECK(hipMemcpy(dB, B.data(), 64 * 8, hipMemcpyDefault));
  double ref[16][16];
  for (int i = 0; i < 16; ++i)
    for (int j = 0; j < 16; ++j) {
      ref[i][j] = 0.0;
      for (int k = 0; k < 4; ++k) ref[i][j] += A[i * 4 + k] * B[k * 16 + j];
    }
  for (int amap = 0; amap < 2; ++amap)
    for (int bmap = 0; bmap < 2; ++bmap) {
      hipLaunchKernelGGL(map_probe, dim3(1), dim3(64), 0, 0, dA, dB, dO,
                         amap, bmap);
      HIP_CHECK(hipDeviceSynchronize());
      std::vector<double> O(256);
      HIP_CHECK(hipMemcpy(O.data(), dO, 256 * 8, hipMemcpyDefault));
      for (int dmap = 0; dmap < 2; ++dmap) {
        double err = 0.0;
        for (int l = 0; l < 64; ++l)
          for (int r = 0; r < 4; ++r) {
            int j = l & 15;
            int i = dmap ? ((l >> 4) + 4 * r) : (4 * (l >> 4) + r);
            err = std::max(err, std::fabs(O[l * 4 + r] - ref[i][j]));
          }
        printf("probe amap=%d bmap=%d dmap=%d err=%.3e%s\n", amap, bmap,
               dmap, err, err < 1e-12 ? "  <-- MATCH" : "");
      }
    }
  HIP_CHECK(hipFree(dA));
  HIP_CHECK(hipFree(dB));
  HIP_CHECK(hipFree(dO));
}

// ------------------------------------------------------------- MFMA peak
__global__ __launch_bounds__(256) void mfma_peak(double* out, int iters) {
  // 4 independent accumulator chains hide the dependent-accumulator
  // latency (64 cyc dep vs 32 cyc issue for the f64 form)
  d4 a0 = {0, 0, 0, 0}, a1 = {0, 0, 0, 0}, a2 = {0, 0, 0, 0}, a3 = {0, 0, 0, 0};
  double a = 1.0 + threadIdx.x * 1e-9;
  double b = 1.0 - threadIdx.x * 1e-9;
  for (int i = 0; i < iters; ++i) {
    a0 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, a0, 0, 0, 0);
    a1 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, a1, 0, 0, 0);
    a2 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, a2, 0, 0, 0);
    a3 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, a3, 0, 0, 0);
  }
  if (a0[0] == 12345.0)
    out[threadIdx.x] = a0[0] + a1[1] + a2[2] + a3[3];
}

// ------------------------------------------------------------------ host
struct Csr {
  std::vector<long> indptr;
  std::vector<int> indices;
  std::vector<double> vals;
  long m, n;
};

struct Bsr {
  std::vector<long> bptr;
  std::vector<int> bcol;
  std::vector<double> bvals;  // 256 per block, row-major
  long nbrows;
};

// 5-diagonal band: offsets {-g, -1, 0, 1, g}
Csr make_band(long m, long g) {
  Csr A;
  A.m = A.n = m;
  A.indptr.assign(m + 1, 0);
  long offs[5] = {-g, -1, 0, 1, g};
  for (long r = 0; r < m; ++r) {
    A.indptr[r + 1] = A.indptr[r];
    for (long d = 0; d < 5; ++d) {
      long c = r + offs[d];
      if (c >= 0 && c < m) {
        A.indices.push_back((int)c);
        A.vals.push_back(d == 2 ? 4.0 : -1.0 + 1e-7 * (double)(r % 13));
        A.indptr[r + 1]++;
      }
    }
  }
  return A;
}

// CSR -> 16-block BSR (blocks padded dense)
Bsr to_bsr(const Csr& A) {
  Bsr B;
  long nb = (A.m + 15) / 16;
  B.nbrows = nb;
  B.bptr.assign(nb + 1, 0);
  for (long br = 0; br < nb; ++br) {
    B.bptr[br + 1] = B.bptr[br];
    std::vector<int> cols;
    for (long r = br * 16; r < std::min(A.m, (br + 1) * 16); ++r)
      for (long p = A.indptr[r]; p < A.indptr[r + 1]; ++p) {
        int bc = A.indices[p] / 16;
        bool seen = false;
        for (int c : cols) seen = seen || (c == bc);
        if (!seen) cols.push_back(bc);
      }
    std::sort(cols.begin(), cols.end());
    for (int bc : cols) {
      B.bcol.push_back(bc);
      size_t base = B.bvals.size();
      B.bvals.resize(base + 256, 0.0);
      for (long r = br * 16; r < std::min(A.m, (br + 1) * 16); ++r)
        for (long p = A.indptr[r]; p < A.indptr[r + 1]; ++p)
          if (A.indices[p] / 16 == bc)
            B.bvals[base + (r - br * 16) * 16 + (A.indices[p] % 16)] =
                A.vals[p];
      B.bptr[br + 1]++;
    }
  }
  return B;
}

// dense-block band: 3 block diagonals, every block fully dense
Csr make_block_dense(long m) {
  Csr A;
  A.m = A.n = m;
  A.indptr.assign(m + 1, 0);
  long nb = m / 16;
  for (long r = 0; r < m; ++r) {
    A.indptr[r + 1] = A.indptr[r];
    long br = r / 16;
    for (long db = -1; db <= 1; ++db) {
      long bc = br + db;
      if (bc < 0 || bc >= nb) continue;
      for (long c = bc * 16; c < bc * 16 + 16; ++c) {
        A.indices.push_back((int)c);
        A.vals.push_back(0.01 * (double)((r * 7 + c * 3) % 11) - 0.05);
        A.indptr[r + 1]++;
      }
    }
  }
  return A;
}

double time_kernel(void (*launch)(void*), void* arg, int iters) {
  launch(arg);  // warm
  HIP_CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  HIP_CHECK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i) launch(arg);
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  return ms / iters;
}

struct Args {
  long m, k, nbrows;
  long *indptr, *bptr;
  int *indices, *bcol;
  double *vals, *bvals, *B, *C;
};

void launch_csr(void* p) {
  Args* a = (Args*)p;
  dim3 block(64, 4), grid((a->k + 63) / 64, (a->m + 3) / 4);
  hipLaunchKernelGGL(csr_spmm, grid, block, 0, 0, a->indptr, a->indices,
                     a->vals, a->B, a->C, a->m, a->k);
}

void launch_bsr(void* p) {
  Args* a = (Args*)p;
  long waves = a->nbrows * (a->k / 16);
  hipLaunchKernelGGL(bsr_mfma_spmm, dim3((waves * 64 + 255) / 256), dim3(256),
                     0, 0, a->bptr, a->bcol, a->bvals, a->B, a->C, a->nbrows,
                     a->k);
}

int check(const Csr& A, const std::vector<double>& B,
          const std::vector<double>& C, long k, const char* tag) {
  double maxerr = 0.0;
  for (long r = 0; r < A.m; r += std::max(1L, A.m / 977)) {
    for (long j = 0; j < k; ++j) {
      double acc = 0.0;
      for (long p = A.indptr[r]; p < A.indptr[r + 1]; ++p)
        acc += A.vals[p] * B[A.indices[p] * k + j];
      maxerr = std::max(maxerr, std::fabs(acc - C[r * k + j]));
    }
  }
  printf("  %s self-check max err %.3e %s\n", tag, maxerr,
         maxerr < 1e-9 ? "OK" : "FAIL");
  return maxerr < 1e-9 ? 0 : 1;
}

int run_case(const char* name, const Csr& A, long k, int iters) {
  Bsr Bs = to_bsr(A);
  long nnz = (long)A.vals.size();
  long bnnz = (long)Bs.bcol.size() * 256;
  printf("%s: m=%ld nnz=%ld k=%ld | blocks=%zu fill=%.1f%%\n", name, A.m, nnz,
         k, Bs.bcol.size(), 100.0 * nnz / bnnz);
  std::vector<double> B(A.n * k), C(A.m * k);
  for (size_t i = 0; i < B.size(); ++i) B[i] = 0.001 * (double)(i % 97) - 0.04;
  Args g;
  g.m = A.m;
  g.k = k;
  g.nbrows = Bs.nbrows;
  HIP_CHECK(hipMalloc(&g.indptr, (A.m + 1) * 8));
  HIP_CHECK(hipMalloc(&g.indices, nnz * 4));
  HIP_CHECK(hipMalloc(&g.vals, nnz * 8));
  HIP_CHECK(hipMalloc(&g.bptr, (Bs.nbrows + 1) * 8));
  HIP_CHECK(hipMalloc(&g.bcol, Bs.bcol.size() * 4));
  HIP_CHECK(hipMalloc(&g.bvals, Bs.bvals.size() * 8));
  HIP_CHECK(hipMalloc(&g.B, B.size() * 8));
  HIP_CHECK(hipMalloc(&g.C, C.size() * 8));
  HIP_CHECK(hipMemcpy(g.indptr, A.indptr.data(), (A.m + 1) * 8, hipMemcpyDefault));
  HIP_CHECK(hipMemcpy(g.indices, A.indices.data(), nnz * 4, hipMemcpyDefault));
  HIP_CHECK(hipMemcpy(g.vals, A.vals.data(), nnz * 8, hipMemcpyDefault));
  HIP_CHECK(hipMemcpy(g.bptr, Bs.bptr.data(), (Bs.nbrows + 1) * 8, hipMemcpyDefault));
  HIP_CHECK(hipMemcpy(g.bcol, Bs.bcol.data(), Bs.bcol.size() * 4, hipMemcpyDefault));
  HIP_CHECK(hipMemcpy(g.bvals, Bs.bvals.data(), Bs.bvals.size() * 8, hipMemcpyDefault));
  HIP_CHECK(hipMemcpy(g.B, B.data(), B.size() * 8, hipMemcpyDefault));

  int rc = 0;
  launch_csr(&g);
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipMemcpy(C.data(), g.C, C.size() * 8, hipMemcpyDefault));
  rc |= check(A, B, C, k, "csr ");
  HIP_CHECK(hipMemset(g.C, 0, C.size() * 8));
  launch_bsr(&g);
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipMemcpy(C.data(), g.C, C.size() * 8, hipMemcpyDefault));
  rc |= check(A, B, C, k, "mfma");

  double ms_csr = time_kernel(launch_csr, &g, iters);
  double ms_bsr = time_kernel(launch_bsr, &g, iters);
  double fl = 2.0 * nnz * k;
  printf("  lane-tiled CSR : %8.3f ms  %8.1f GFLOP/s (nnz flops)\n", ms_csr,
         fl / ms_csr / 1e6);
  printf("  MFMA 16x16 BSR : %8.3f ms  %8.1f GFLOP/s (nnz flops; block "
         "flops %.1f)\n",
         ms_bsr, fl / ms_bsr / 1e6, 2.0 * bnnz * k / ms_bsr / 1e6);
  HIP_CHECK(hipFree(g.indptr));
  HIP_CHECK(hipFree(g.indices));
  HIP_CHECK(hipFree(g.vals));
  HIP_CHECK(hipFree(g.bptr));
  HIP_CHECK(hipFree(g.bcol));
  HIP_CHECK(hipFree(g.bvals));
  HIP_CHECK(hipFree(g.B));
  HIP_CHECK(hipFree(g.C));
  return rc;
}

int main(int argc, char** argv) {
  if (argc > 1 && std::string(argv[1]) == "--probe") {
    probe_mapping();
    probe_mapping_f32();
    return 0;
  }
  long m = argc > 1 ? atol(argv[1]) : (1L << 20);
  m = (m / 16) * 16;
  int iters = argc > 2 ? atoi(argv[2]) : 20;

  // MFMA peak probe: 256 threads/block, many blocks, register-resident
  {
    double* out;
    HIP_CHECK(hipMalloc(&out, 256 * 8));
    int it = 20000;
    hipLaunchKernelGGL(mfma_peak, dim3(4096), dim3(256), 0, 0, out, it);
    HIP_CHECK(hipDeviceSynchronize());
    hipEvent_t t0, t1;
    HIP_CHECK(hipEventCreate(&t0));
    HIP_CHECK(hipEventCreate(&t1));
    HIP_CHECK(hipEventRecord(t0));
    hipLaunchKernelGGL(mfma_peak, dim3(4096), dim3(256), 0, 0, out, it);
    HIP_CHECK(hipEventRecord(t1));
    HIP_CHECK(hipEventSynchronize(t1));
    float ms;
    HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
    double waves = 4096.0 * 256 / 64;
    double flops = waves * 4.0 * it * 2.0 * 16 * 16 * 4;  // 2*M*N*K per mfma
    printf("mfma_f64_16x16x4 peak probe: %.1f GFLOP/s fp64\n",
           flops / ms / 1e6);
    HIP_CHECK(hipFree(out));
  }

  int rc = 0;
  long g = (long)std::sqrt((double)m);
  for (long k : {32L, 64L}) {
    Csr band = make_band(m, g);
    rc |= run_case("band5 (poisson-like)", band, k, iters);
    Csr bd = make_block_dense(m / 4);
    rc |= run_case("block-dense 16x16", bd, k, iters);
  }
  printf(rc ? "SELF-CHECK FAILED\n" : "ALL OK\n");
  return rc;
}
