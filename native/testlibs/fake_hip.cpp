// Fake libamdhip64 — the CPU loopback substrate (SURVEY.md §7 phase 2:
// "fake HIP symbols so the token protocol is unit-testable on CPU",
// ranked among the hardest parts the reference never had).
//
// Built as libamdhip64.so.7 under native/testlibs/ ONLY; test drivers
// link it explicitly via -L/rpath. It never shadows the real runtime:
// nothing outside tests/test_hook_loopback.py puts this directory on a
// library path.
//
// Behavior: counts calls, "allocates" with malloc, models GPU work as
// an accumulated virtual backlog that hipDeviceSynchronize "drains" by
// sleeping, so the hook's drain-at-renewal and wall-accounting paths
// run exactly as on hardware.
#define __HIP_PLATFORM_AMD__ 1
#include <hip/hip_runtime_api.h>

#include <atomic>
#include <cstdlib>
#include <cstring>
#include <ctime>

static std::atomic<long long> g_launches{0};
static std::atomic<long long> g_memcpys{0};
static std::atomic<long long> g_syncs{0};
static std::atomic<long long> g_allocated{0};
static std::atomic<long long> g_backlog_us{0};  // queued "GPU work"

extern "C" {

// introspection for the test driver
long long fake_hip_launches(void) { return g_launches.load(); }
long long fake_hip_syncs(void) { return g_syncs.load(); }
long long fake_hip_allocated(void) { return g_allocated.load(); }
void fake_hip_queue_work_us(long long us) { g_backlog_us += us; }

hipError_t hipMalloc(void** ptr, size_t size) {
  *ptr = malloc(size ? size : 1);
  g_allocated += (long long)size;
  return *ptr ? hipSuccess : hipErrorOutOfMemory;
}

hipError_t hipFree(void* ptr) {
  free(ptr);
  return hipSuccess;
}

hipError_t hipMallocManaged(void** ptr, size_t size, unsigned int) {
  return hipMalloc(ptr, size);
}

// VMM family (PyTorch expandable_segments path) + pitched allocs — the
// cap-bypass surfaces closed in round 2 (hiphook.cpp VMM section)
hipError_t hipMemCreate(hipMemGenericAllocationHandle_t* handle, size_t size,
                        const hipMemAllocationProp*, unsigned long long) {
  *handle = (hipMemGenericAllocationHandle_t)malloc(16);
  g_allocated += (long long)size;
  return *handle ? hipSuccess : hipErrorOutOfMemory;
}

hipError_t hipMemRelease(hipMemGenericAllocationHandle_t handle) {
  free((void*)handle);
  return hipSuccess;
}

hipError_t hipMallocPitch(void** ptr, size_t* pitch, size_t width,
                          size_t height) {
  *pitch = (width + 255) / 256 * 256;  // model the HW pitch padding
  size_t bytes = *pitch * height;
  *ptr = malloc(bytes ? bytes : 1);
  g_allocated += (long long)bytes;
  return *ptr ? hipSuccess : hipErrorOutOfMemory;
}

hipError_t hipMemGetInfo(size_t* free_b, size_t* total_b) {
  if (total_b) *total_b = 16ull << 30;
  if (free_b) *free_b = (16ull << 30) - (size_t)g_allocated.load();
  return hipSuccess;
}

hipError_t hipLaunchKernel(const void*, dim3, dim3, void**, size_t,
                           hipStream_t) {
  g_launches++;
  return hipSuccess;
}

hipError_t hipMemcpyAsync(void*, const void*, size_t, hipMemcpyKind,
                          hipStream_t) {
  g_memcpys++;
  return hipSuccess;
}

hipError_t hipStreamIsCapturing(hipStream_t, hipStreamCaptureStatus* st) {
  if (st) *st = hipStreamCaptureStatusNone;
  return hipSuccess;
}

hipError_t hipDeviceSynchronize(void) {
  g_syncs++;
  long long us = g_backlog_us.exchange(0);
  if (us > 0) {
    struct timespec ts = {us / 1000000, (us % 1000000) * 1000};
    nanosleep(&ts, nullptr);
  }
  return hipSuccess;
}

}  // extern "C"
