// hook_app — a fake "GPU application" linked against the FAKE
// libamdhip64 and run with LD_PRELOAD=libhiphook.so: exercises the real
// interposer end-to-end on a CPU-only box (tests/test_hook_loopback.py).
//
// Scenarios (selected by argv[1]):
//   memcap   KUBESHARE_GPU_MEM enforcement + hipMemGetInfo clamping
//   gate     token leases against a live gpu-schd while "computing"
//   rccl     launches from librccl.so.1 bypass the gate
#define __HIP_PLATFORM_AMD__ 1
#include <dlfcn.h>
#include <hip/hip_runtime_api.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <time.h>
#include <unistd.h>

extern "C" {
long long fake_hip_launches(void);
long long fake_hip_syncs(void);
void fake_hip_queue_work_us(long long us);
// injected by the LD_PRELOADed hook; weak so the app links without it
long long ks_hook_leases(void) __attribute__((weak));
long long ks_hook_mem_denied(void) __attribute__((weak));
int ks_hook_active(void) __attribute__((weak));
}

static double now_ms() {
  struct timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return ts.tv_sec * 1e3 + ts.tv_nsec * 1e-6;
}

static int scenario_memcap() {
  // cap set by the test to 1 GiB
  void* a = nullptr;
  if (hipMalloc(&a, 512ull << 20) != hipSuccess) return 10;
  void* b = nullptr;
  if (hipMalloc(&b, 768ull << 20) != hipErrorOutOfMemory) return 11;
  if (!ks_hook_mem_denied || ks_hook_mem_denied() != 1) return 12;
  size_t free_b = 0, total_b = 0;
  if (hipMemGetInfo(&free_b, &total_b) != hipSuccess) return 13;
  if (total_b != (1ull << 30)) return 14;           // clamped to the cap
  if (free_b != (1ull << 30) - (512ull << 20)) return 15;
  if (hipFree(a) != hipSuccess) return 16;
  if (hipMalloc(&b, 768ull << 20) != hipSuccess) return 17;  // fits again
  printf("MEMCAP_OK\n");
  return 0;
}

static int scenario_gate(double duration_ms) {
  if (!ks_hook_active || ks_hook_active() != 1) return 20;
  dim3 d{1, 1, 1};
  double t0 = now_ms();
  long long launches = 0;
  while (now_ms() - t0 < duration_ms) {
    // each "kernel" queues 2 ms of virtual GPU work; the hook's renewal
    // drain (hipDeviceSynchronize on the fake) sleeps it off
    fake_hip_queue_work_us(2000);
    if (hipLaunchKernel(nullptr, d, d, nullptr, 0, nullptr) != hipSuccess)
      return 21;
    launches++;
    usleep(2000);
  }
  printf("GATE_OK launches=%lld leases=%lld syncs=%lld\n", launches,
         ks_hook_leases ? ks_hook_leases() : -1, fake_hip_syncs());
  return 0;
}

static int scenario_rccl() {
  if (!ks_hook_active || ks_hook_active() != 1) return 30;
  void* h = dlopen("librccl.so.1", RTLD_NOW | RTLD_LOCAL);
  if (!h) {
    fprintf(stderr, "dlopen librccl: %s\n", dlerror());
    return 31;
  }
  auto fn = (hipError_t (*)(int))dlsym(h, "fake_rccl_allreduce");
  if (!fn) return 32;
  long long leases_before = ks_hook_leases ? ks_hook_leases() : 0;
  if (fn(50) != hipSuccess) return 33;  // 50 collective launches
  long long leases_after = ks_hook_leases ? ks_hook_leases() : 0;
  if (fake_hip_launches() < 50) return 34;      // they really ran
  if (leases_after != leases_before) return 35;  // and were NOT gated
  printf("RCCL_OK\n");
  return 0;
}

static int scenario_vmm() {
  // cap set by the test to 1 GiB; the VMM + pitched paths must honor it
  hipMemGenericAllocationHandle_t h1 = nullptr, h2 = nullptr;
  if (hipMemCreate(&h1, 512ull << 20, nullptr, 0) != hipSuccess) return 40;
  if (hipMemCreate(&h2, 768ull << 20, nullptr, 0) != hipErrorOutOfMemory)
    return 41;  // over cap: physical commit refused
  // pitched alloc: 256 MiB worth of rows also counts against the cap
  void* p = nullptr;
  size_t pitch = 0;
  if (hipMallocPitch(&p, &pitch, 1 << 20, 256) != hipSuccess) return 42;
  if (pitch * 256 < (256ull << 20)) return 43;
  // 512 + 256 = 768 MiB used; another 512 MiB VMM chunk must fail...
  if (hipMemCreate(&h2, 512ull << 20, nullptr, 0) != hipErrorOutOfMemory)
    return 44;
  // ...until the first handle is released
  if (hipMemRelease(h1) != hipSuccess) return 45;
  if (hipMemCreate(&h2, 512ull << 20, nullptr, 0) != hipSuccess) return 46;
  printf("VMM_OK\n");
  return 0;
}

int main(int argc, char** argv) {
  if (argc < 2) return 2;
  if (strcmp(argv[1], "memcap") == 0) return scenario_memcap();
  if (strcmp(argv[1], "vmm") == 0) return scenario_vmm();
  if (strcmp(argv[1], "gate") == 0)
    return scenario_gate(argc >= 3 ? atof(argv[2]) : 2000.0);
  if (strcmp(argv[1], "rccl") == 0) return scenario_rccl();
  return 2;
}
