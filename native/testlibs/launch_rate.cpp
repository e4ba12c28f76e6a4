// launch_rate — dispatch hot-path microbench (CPU, fake-HIP substrate).
// Measures ns/hipLaunchKernel with and without the LD_PRELOAD gate
// (tools/hook_overhead.py orchestrates; numbers in
// profiles/hook_hotpath_overhead.txt).
#define __HIP_PLATFORM_AMD__ 1
#include <hip/hip_runtime_api.h>
#include <stdio.h>
#include <time.h>

static double now_ms() {
  struct timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return ts.tv_sec * 1e3 + ts.tv_nsec * 1e-6;
}

int main() {
  dim3 d{1, 1, 1};
  for (int i = 0; i < 10000; i++)
    (void)hipLaunchKernel(nullptr, d, d, nullptr, 0, nullptr);
  double t0 = now_ms();
  const long N = 5000000;
  for (long i = 0; i < N; i++)
    (void)hipLaunchKernel(nullptr, d, d, nullptr, 0, nullptr);
  double dt = now_ms() - t0;
  printf("%.1f ns/launch (%ld launches in %.1f ms)\n", dt * 1e6 / N, N, dt);
  return 0;
}
