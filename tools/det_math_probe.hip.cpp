// Debug probe: dump device det_gaussian intermediates for given (u1,u2)
// pairs so they can be diffed bit-for-bit against utils/det_math.py.
// Build ON a GPU box:
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/det_math_probe.hip.cpp \
//         -o gpurun_out/det_probe -I isolation_forest_amd/ops/hip
// Run: ./gpurun_out/det_probe < pairs.txt  (lines: "u1 u2")
// Output lines: u1 u2 logbits cosbits gbits (hex u64)

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <vector>

#include "det_math.h"

__global__ void probe(const double* u1, const double* u2, double* logv,
                      double* cosv, double* g, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  double t = 1.0 - u1[i];
  if (t < 1e-300) t = 1e-300;
  logv[i] = ifa::det_log(t);
  cosv[i] = ifa::det_cos2pi(u2[i]);
  g[i] = ifa::det_gaussian(u1[i], u2[i]);
}

int main() {
  std::vector<double> a, b;
  double x, y;
  while (scanf("%lf %lf", &x, &y) == 2) {
    a.push_back(x);
    b.push_back(y);
  }
  int n = (int)a.size();
  double *da, *db, *dl, *dc, *dg;
  hipMalloc(&da, n * 8);
  hipMalloc(&db, n * 8);
  hipMalloc(&dl, n * 8);
  hipMalloc(&dc, n * 8);
  hipMalloc(&dg, n * 8);
  hipMemcpy(da, a.data(), n * 8, hipMemcpyHostToDevice);
  hipMemcpy(db, b.data(), n * 8, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(probe, dim3((n + 255) / 256), dim3(256), 0, 0, da, db, dl,
                     dc, dg, n);
  std::vector<double> l(n), c(n), g(n);
  hipMemcpy(l.data(), dl, n * 8, hipMemcpyDeviceToHost);
  hipMemcpy(c.data(), dc, n * 8, hipMemcpyDeviceToHost);
  hipMemcpy(g.data(), dg, n * 8, hipMemcpyDeviceToHost);
  for (int i = 0; i < n; ++i) {
    printf("%.17g %.17g %016llx %016llx %016llx\n", a[i], b[i],
           (unsigned long long)__builtin_bit_cast(uint64_t, l[i]),
           (unsigned long long)__builtin_bit_cast(uint64_t, c[i]),
           (unsigned long long)__builtin_bit_cast(uint64_t, g[i]));
  }
  return 0;
}
