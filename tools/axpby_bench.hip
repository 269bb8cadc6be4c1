// axpby variant A/B: y = y + s*x over 268M fp64 (the CG vector-op shape)
#include <hip/hip_runtime.h>
#include <algorithm>
#include <cstdio>
#include <vector>
#define CHECK(c) do { hipError_t e=(c); if(e!=hipSuccess){printf("ERR %s %d\n", hipGetErrorString(e), __LINE__); exit(1);} } while(0)

struct D2 { double a, b; };
struct D4 { double a, b, c, d; };

__global__ __launch_bounds__(256) void v_stride2(double* y, const double* x, double s, int64_t half) {
  int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x;
  int64_t st = (int64_t)gridDim.x * 256;
  auto* y2 = reinterpret_cast<D2*>(y); auto* x2 = reinterpret_cast<const D2*>(x);
  for (; i < half; i += st) { D2 yv=y2[i]; D2 xv=x2[i]; yv.a+=s*xv.a; yv.b+=s*xv.b; y2[i]=yv; }
}
__global__ __launch_bounds__(256) void v_flat2(double* y, const double* x, double s, int64_t half) {
  int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x;
  if (i >= half) return;
  auto* y2 = reinterpret_cast<D2*>(y); auto* x2 = reinterpret_cast<const D2*>(x);
  D2 yv=y2[i]; D2 xv=x2[i]; yv.a+=s*xv.a; yv.b+=s*xv.b; y2[i]=yv;
}
__global__ __launch_bounds__(256) void v_flat4(double* y, const double* x, double s, int64_t quarter) {
  int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x;
  if (i >= quarter) return;
  auto* y4 = reinterpret_cast<D4*>(y); auto* x4 = reinterpret_cast<const D4*>(x);
  D4 yv=y4[i]; D4 xv=x4[i]; yv.a+=s*xv.a; yv.b+=s*xv.b; yv.c+=s*xv.c; yv.d+=s*xv.d; y4[i]=yv;
}
__global__ __launch_bounds__(256) void v_stride4(double* y, const double* x, double s, int64_t quarter) {
  int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x;
  int64_t st = (int64_t)gridDim.x * 256;
  auto* y4 = reinterpret_cast<D4*>(y); auto* x4 = reinterpret_cast<const D4*>(x);
  for (; i < quarter; i += st) { D4 yv=y4[i]; D4 xv=x4[i]; yv.a+=s*xv.a; yv.b+=s*xv.b; yv.c+=s*xv.c; yv.d+=s*xv.d; y4[i]=yv; }
}

int main() {
  int64_t n = 268435456;
  double *x, *y;
  CHECK(hipMalloc(&x, n*8)); CHECK(hipMalloc(&y, n*8));
  CHECK(hipMemset(x, 0, n*8)); CHECK(hipMemset(y, 0, n*8));
  hipEvent_t t0, t1; hipEventCreate(&t0); hipEventCreate(&t1);
  struct V { const char* name; int id; };
  std::vector<V> vs = {{"stride2_32k",0},{"flat2",1},{"flat4",2},{"stride4_32k",3},{"stride2_4k",4},{"stride4_2k",5}};
  std::vector<std::vector<float>> times(vs.size());
  for (int round = 0; round < 7; ++round)
    for (size_t vi = 0; vi < vs.size(); ++vi) {
      hipEventRecord(t0);
      for (int rep = 0; rep < 3; ++rep) {
        switch (vs[vi].id) {
          case 0: hipLaunchKernelGGL(v_stride2, dim3(32768), dim3(256), 0, 0, y, x, 1.5, n/2); break;
          case 1: hipLaunchKernelGGL(v_flat2, dim3((n/2+255)/256), dim3(256), 0, 0, y, x, 1.5, n/2); break;
          case 2: hipLaunchKernelGGL(v_flat4, dim3((n/4+255)/256), dim3(256), 0, 0, y, x, 1.5, n/4); break;
          case 3: hipLaunchKernelGGL(v_stride4, dim3(32768), dim3(256), 0, 0, y, x, 1.5, n/4); break;
          case 4: hipLaunchKernelGGL(v_stride2, dim3(4096), dim3(256), 0, 0, y, x, 1.5, n/2); break;
          case 5: hipLaunchKernelGGL(v_stride4, dim3(2048), dim3(256), 0, 0, y, x, 1.5, n/4); break;
        }
      }
      hipEventRecord(t1); CHECK(hipEventSynchronize(t1));
      float ms; hipEventElapsedTime(&ms, t0, t1); times[vi].push_back(ms/3);
    }
  for (size_t vi = 0; vi < vs.size(); ++vi) {
    std::sort(times[vi].begin(), times[vi].end());
    float med = times[vi][times[vi].size()/2];
    printf("%-12s %7.3f ms  %7.1f GB/s\n", vs[vi].name, med, 3.0*n*8/(med*1e6));
  }
  return 0;
}
