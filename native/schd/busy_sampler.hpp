// Server-side GPU-busy sampler for gpu-schd (lease accounting,
// MI355X-native).
//
// Charges each lease the GPU time it actually used instead of wall
// time: gpu-schd polls the device's busy percent (amdgpu GRBM activity
// via librocm_smi64's rsmi_dev_busy_percent_get) every few ms and
// credits busy_ms = dt * busy% to the current token holder
// (TokenScheduler::add_busy). Advantages over client-side hipEvent
// accounting (the alternative considered in round 1):
//   - zero per-dispatch overhead in the hook's hot path (an event
//     record per gated launch would cost ~2 us x thousands of
//     launches/step);
//   - a CPU-bound phase inside a lease (dataloader stall, sparse
//     dispatch) is NOT charged — quota error stays low for bursty
//     pods, not just saturating ones;
//   - measured on the host, outside the container: a hostile hook
//     cannot under-report.
//
// librocm_smi64 is dlopen'd so gpu-schd still builds and runs on
// CPU-only boxes (tests); without the library the scheduler keeps the
// round-1 wall/RET accounting.
#pragma once

#include <dlfcn.h>
#include <stdint.h>

#include "../common/protocol.hpp"

namespace ks {

class BusySampler {
 public:
  // device_index: rocm-smi enumeration order (the launcher passes the
  // same index it used for ROCR_VISIBLE_DEVICES / the sched port).
  bool init(int device_index) {
    dev_ = (uint32_t)device_index;
    lib_ = dlopen("librocm_smi64.so.7", RTLD_NOW | RTLD_LOCAL);
    if (!lib_) lib_ = dlopen("librocm_smi64.so", RTLD_NOW | RTLD_LOCAL);
    if (!lib_) return false;
    auto init_fn =
        (int (*)(uint64_t))dlsym(lib_, "rsmi_init");
    busy_get_ =
        (int (*)(uint32_t, uint32_t*))dlsym(lib_, "rsmi_dev_busy_percent_get");
    if (!init_fn || !busy_get_ || init_fn(0) != 0) {
      busy_get_ = nullptr;
      return false;
    }
    uint32_t pct = 0;
    if (busy_get_(dev_, &pct) != 0) {  // device not visible/supported
      busy_get_ = nullptr;
      return false;
    }
    last_t_ = now_ms();
    return true;
  }

  bool active() const { return busy_get_ != nullptr; }

  // Busy milliseconds accumulated since the previous poll (dt * busy%).
  double poll() {
    if (!busy_get_) return 0.0;
    double now = now_ms();
    double dt = now - last_t_;
    last_t_ = now;
    if (dt <= 0.0) return 0.0;
    uint32_t pct = 0;
    if (busy_get_(dev_, &pct) != 0) return 0.0;
    if (pct > 100) pct = 100;
    return dt * (double)pct / 100.0;
  }

 private:
  void* lib_ = nullptr;
  uint32_t dev_ = 0;
  double last_t_ = 0.0;
  int (*busy_get_)(uint32_t, uint32_t*) = nullptr;
};

}  // namespace ks
