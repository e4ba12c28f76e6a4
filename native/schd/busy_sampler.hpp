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
  // rsmi_utilization_counter_t: { RSMI_UTILIZATION_COUNTER_TYPE type;
  // uint64_t value; } — enum + padding puts value at offset 8.
  struct UtilCounter {
    int32_t type;      // 0 = RSMI_COARSE_GRAIN_GFX_ACTIVITY
    int32_t pad_;
    uint64_t value;
  };
  static_assert(sizeof(UtilCounter) == 16, "rsmi ABI");

  // device_index: rocm-smi enumeration order (the launcher passes the
  // same index it used for ROCR_VISIBLE_DEVICES / the sched port).
  bool init(int device_index) {
    dev_ = (uint32_t)device_index;
    lib_ = dlopen("librocm_smi64.so.7", RTLD_NOW | RTLD_LOCAL);
    if (!lib_) lib_ = dlopen("librocm_smi64.so", RTLD_NOW | RTLD_LOCAL);
    if (!lib_) return false;
    auto init_fn = (int (*)(uint64_t))dlsym(lib_, "rsmi_init");
    if (!init_fn || init_fn(0) != 0) return false;
    // Preferred: the accumulating coarse-grain GFX activity counter —
    // "every millisecond the firmware calculates % busy and accumulates
    // it" (rocm_smi.h) — so busy_ms over any interval = delta/100,
    // exact even for sub-sampling-window bursts. Fallback: polling the
    // instantaneous (SMU-averaged) busy percent, which dilutes bursts.
    count_get_ = (int (*)(uint32_t, UtilCounter*, uint32_t, uint64_t*))
        dlsym(lib_, "rsmi_utilization_count_get");
    if (count_get_) {
      UtilCounter c{0, 0, 0};
      uint64_t ts = 0;
      if (count_get_(dev_, &c, 1, &ts) == 0) {
        last_acc_ = c.value;
      } else {
        count_get_ = nullptr;
      }
    }
    busy_get_ = (int (*)(uint32_t, uint32_t*))
        dlsym(lib_, "rsmi_dev_busy_percent_get");
    if (!count_get_) {
      uint32_t pct = 0;
      if (!busy_get_ || busy_get_(dev_, &pct) != 0) {
        busy_get_ = nullptr;
        return false;
      }
    }
    last_t_ = now_ms();
    return true;
  }

  bool active() const { return count_get_ != nullptr || busy_get_ != nullptr; }
  bool accumulating() const { return count_get_ != nullptr; }

  // Busy milliseconds since the previous poll.
  double poll() {
    if (count_get_) {
      UtilCounter c{0, 0, 0};
      uint64_t ts = 0;
      if (count_get_(dev_, &c, 1, &ts) == 0 && c.value >= last_acc_) {
        double busy = (double)(c.value - last_acc_) / 100.0;
        last_acc_ = c.value;
        return busy;
      }
      return 0.0;
    }
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
  uint64_t last_acc_ = 0;
  int (*busy_get_)(uint32_t, uint32_t*) = nullptr;
  int (*count_get_)(uint32_t, UtilCounter*, uint32_t, uint64_t*) = nullptr;
};

}  // namespace ks
