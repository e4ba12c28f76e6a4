// Fake librccl — built as librccl.so.1 under native/testlibs/. Its one
// export launches a "collective kernel" through the (interposed) HIP
// runtime, so tests can verify the hook's caller-library exemption:
// launches whose call site lives in librccl must bypass the token gate.
#define __HIP_PLATFORM_AMD__ 1
#include <hip/hip_runtime_api.h>

extern "C" hipError_t fake_rccl_allreduce(int iters) {
  dim3 d{1, 1, 1};
  for (int i = 0; i < iters; i++) {
    hipError_t e = hipLaunchKernel(nullptr, d, d, nullptr, 0, nullptr);
    if (e != hipSuccess) return e;
  }
  return hipSuccess;
}
