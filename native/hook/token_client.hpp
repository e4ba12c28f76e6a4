// Token-gate client used by libhiphook.so — pure protocol logic, no HIP
// dependency, so the whole gate state machine is unit-testable on a
// CPU-only box (hook_selftest.cpp and tests/test_native_daemons.py).
//
// Gate semantics (see DESIGN.md and native/schd/token_sched.hpp): the
// pod holds at most one wall-time lease; between leases every gated HIP
// call blocks in acquire(). The caller (hiphook.cpp) drains the GPU
// before return_token() so the wall measurement equals GPU-busy time
// for a GPU-bound pod.
#pragma once

#include <pthread.h>

#include <atomic>
#include <string>

#include "../common/protocol.hpp"

namespace ks {

class TokenClient {
 public:
  // endpoint: "ip:port" handled by caller — we take host (or uds path)
  // + port. pod is "<namespace>/<name>" (pod-mgr re-stamps it anyway).
  void configure(const std::string& host, int port, const std::string& pod) {
    host_ = host;
    port_ = port;
    pod_ = pod.empty() ? "unknown/unknown" : pod;
  }

  bool configured() const { return !host_.empty(); }

  // Blocking: REQ ... wait for GRANT. Returns quota_ms (<=0 on failure —
  // caller decides whether to fail open or hard).
  double acquire(double hint_ms) {
    int fd = ensure_fd();
    if (fd < 0) return -1.0;
    char line[256];
    snprintf(line, sizeof(line), "REQ %s %.3f", pod_.c_str(), hint_ms);
    if (!send_line(fd, line)) return reconnect_fail();
    std::string reply;
    // GRANT may be delayed arbitrarily (that IS the throttle) — but OK
    // replies to a pipelined RET may arrive first; skip them.
    for (;;) {
      if (!recv_line(fd, reply)) return reconnect_fail();
      if (reply.rfind("GRANT", 0) == 0) break;
      if (reply.rfind("OK", 0) == 0) continue;
      return -1.0;
    }
    requests_++;
    return atof(reply.c_str() + 5);
  }

  // Non-blocking-ish: RET <used>. The OK is consumed lazily by the next
  // acquire() (pipelining keeps the gate off the hot path).
  bool return_token(double used_ms) {
    int fd = ensure_fd();
    if (fd < 0) return false;
    char line[256];
    snprintf(line, sizeof(line), "RET %s %.3f", pod_.c_str(), used_ms);
    if (!send_line(fd, line)) {
      reconnect_fail();
      return false;
    }
    pending_ok_++;
    returned_ms_ += used_ms;
    return true;
  }

  bool report_mem(long long bytes) {
    int fd = ensure_fd();
    if (fd < 0) return false;
    char line[128];
    snprintf(line, sizeof(line), "MEM %s %lld", pod_.c_str(), bytes);
    if (!send_line(fd, line)) {
      reconnect_fail();
      return false;
    }
    pending_ok_++;
    return true;
  }

  // Called in the forked child: the socket fd is shared with the parent,
  // using it from both corrupts the stream — abandon it.
  void reset_after_fork() {
    fd_ = -1;  // deliberately not closed: the parent still owns it
    pending_ok_ = 0;
  }

  long long requests() const { return requests_; }
  double returned_ms() const { return returned_ms_; }

 private:
  int ensure_fd() {
    if (fd_ >= 0) return fd_;
    if (host_.empty()) return -1;
    fd_ = connect_to(host_.c_str(), port_, connect_timeout_ms_);
    if (fd_ >= 0) {
      // drain nothing; fresh stream
      pending_ok_ = 0;
    }
    return fd_;
  }
  double reconnect_fail() {
    if (fd_ >= 0) ::close(fd_);
    fd_ = -1;
    return -1.0;
  }

  std::string host_;
  int port_ = 0;
  std::string pod_;
  int fd_ = -1;
  int pending_ok_ = 0;
  long long requests_ = 0;
  double returned_ms_ = 0.0;
  double connect_timeout_ms_ = 15000.0;
};

// The gate: fast-path check shared by every hooked entry point.
// drain() is injected so this stays HIP-free (hiphook.cpp passes
// hipDeviceSynchronize; tests pass a stub that models GPU work).
//
// Idle release: a lease holder that stops dispatching (end of job,
// CPU-bound phase, waiting on input) must not sit on the token until
// expiry/revocation — that would idle the GPU for everyone. A watchdog
// thread relinquishes the lease once no gated call has arrived for
// idle_release_ms (drains first, so queued work is charged), making the
// slicing truly work-conserving; the next dispatch re-acquires.
class TokenGate {
 public:
  using DrainFn = void (*)(void*);

  void init(const std::string& host, int port, const std::string& pod,
            DrainFn drain, void* drain_arg) {
    client_.configure(host, port, pod);
    drain_ = drain;
    drain_arg_ = drain_arg;
    if (const char* e = getenv("KUBESHARE_IDLE_RELEASE_MS"))
      idle_release_ms_ = atof(e);
    if (const char* e = getenv("KUBESHARE_LEASE_HINT"))
      hint_enabled_ = e[0] != '0';
    enabled_.store(client_.configured(), std::memory_order_release);
  }

  bool enabled() const { return enabled_.load(std::memory_order_acquire); }

  // The dispatch fast path: records activity (for the idle watchdog)
  // and reports whether the current lease covers `now`.
  inline bool touch_lease_valid() {
    double now = now_ms();
    last_activity_.store(now, std::memory_order_relaxed);
    return now < deadline_.load(std::memory_order_acquire);
  }

  // The hot path: ~ns when the lease is valid.
  inline void ensure() {
    if (!enabled()) return;
    double now = now_ms();
    last_activity_.store(now, std::memory_order_relaxed);
    if (now < deadline_.load(std::memory_order_acquire)) return;
    renew();
  }

  // Expire the current lease now (used at teardown).
  void relinquish() {
    pthread_mutex_lock(&mu_);
    if (holding_) {
      if (drain_) drain_(drain_arg_);
      give_back(now_ms() - grant_time_);
      holding_ = false;
      deadline_.store(-1.0, std::memory_order_release);
    }
    pthread_mutex_unlock(&mu_);
  }

  void report_mem(long long bytes) {
    pthread_mutex_lock(&mu_);
    client_.report_mem(bytes);
    pthread_mutex_unlock(&mu_);
  }

  void reset_after_fork() {
    // mutex state may be inconsistent post-fork; re-init. The watchdog
    // thread does not survive fork — restarted on the next acquire.
    pthread_mutex_init(&mu_, nullptr);
    client_.reset_after_fork();
    holding_ = false;
    watchdog_running_.store(false, std::memory_order_release);
    deadline_.store(-1.0, std::memory_order_release);
  }

  long long leases() const { return client_.requests(); }
  double used_ms_total() const { return client_.returned_ms(); }
  double acquire_wait_ms_total() const { return acquire_wait_ms_; }

 private:
  static void* watchdog_entry(void* self) {
    static_cast<TokenGate*>(self)->watchdog_loop();
    return nullptr;
  }

  void watchdog_loop() {
    const double tick = std::max(5.0, idle_release_ms_ / 4.0);
    for (;;) {
      usleep((useconds_t)(tick * 1000));
      if (!enabled()) continue;
      double dl = deadline_.load(std::memory_order_acquire);
      if (dl < 0) continue;  // no lease
      double idle = now_ms() - last_activity_.load(std::memory_order_relaxed);
      if (idle < idle_release_ms_) continue;
      // No dispatches for a while: give the token back (drains first,
      // so still-running queued work is fully charged).
      pthread_mutex_lock(&mu_);
      if (holding_ &&
          now_ms() - last_activity_.load(std::memory_order_relaxed) >=
              idle_release_ms_) {
        double t_pre = now_ms();
        if (drain_) drain_(drain_arg_);
        double t_post = now_ms();
        double used;
        if (t_post - t_pre < 2.0) {
          // GPU was already idle: charge only up to last activity +
          // the idle threshold, not the watchdog's reaction time
          used = std::min(
              t_post,
              last_activity_.load(std::memory_order_relaxed) +
                  idle_release_ms_) - grant_time_;
        } else {
          used = t_post - grant_time_;  // drain did real work
        }
        give_back(std::max(0.0, used));
        holding_ = false;
        deadline_.store(-1.0, std::memory_order_release);
        idle_releases_++;
      }
      pthread_mutex_unlock(&mu_);
    }
  }

  // mu_ held. Reports use upstream and maintains the per-lease EWMA
  // that renew() sends as the REQ hint.
  void give_back(double used_ms) {
    used_ms = std::max(0.0, used_ms);
    ewma_used_ = ewma_used_ <= 0.0 ? used_ms
                                   : 0.7 * ewma_used_ + 0.3 * used_ms;
    client_.return_token(used_ms);
  }

  void start_watchdog_locked() {
    if (watchdog_running_.load(std::memory_order_acquire)) return;
    pthread_t t;
    if (pthread_create(&t, nullptr, watchdog_entry, this) == 0) {
      pthread_detach(t);
      watchdog_running_.store(true, std::memory_order_release);
    }
  }

  void renew() {
    pthread_mutex_lock(&mu_);
    double now = now_ms();
    if (now < deadline_.load(std::memory_order_relaxed)) {
      pthread_mutex_unlock(&mu_);
      return;  // another thread already renewed
    }
    if (holding_) {
      // Drain BEFORE reporting so `used` covers all async work we
      // submitted inside the lease.
      if (drain_) drain_(drain_arg_);
      double used = now_ms() - grant_time_;
      give_back(used);
      holding_ = false;
    }
    start_watchdog_locked();
    double t_wait0 = now_ms();
    // hint = EWMA of recent per-lease use, so gpu-schd can right-size
    // the next quota (a bursty pod gets shorter leases; see
    // token_sched.hpp schedule()). KUBESHARE_LEASE_HINT=0 disables.
    double quota = client_.acquire(hint_enabled_ ? ewma_used_ : 0.0);
    acquire_wait_ms_ += now_ms() - t_wait0;
    if (quota <= 0.0) {
      // Scheduler unreachable: fail OPEN for liveness (the node daemon
      // restarts the chain; isolation degrades, jobs don't die) unless
      // KUBESHARE_REQUIRE_HOOK demands otherwise (checked by caller via
      // failed() — hiphook aborts there).
      failures_++;
      deadline_.store(now_ms() + 1000.0, std::memory_order_release);
      pthread_mutex_unlock(&mu_);
      return;
    }
    grant_time_ = now_ms();
    last_activity_.store(grant_time_, std::memory_order_relaxed);
    deadline_.store(grant_time_ + quota, std::memory_order_release);
    holding_ = true;
    pthread_mutex_unlock(&mu_);
  }

  TokenClient client_;
  DrainFn drain_ = nullptr;
  void* drain_arg_ = nullptr;
  pthread_mutex_t mu_ = PTHREAD_MUTEX_INITIALIZER;
  std::atomic<bool> enabled_{false};
  std::atomic<bool> watchdog_running_{false};
  std::atomic<double> deadline_{-1.0};
  std::atomic<double> last_activity_{0.0};
  double idle_release_ms_ = 25.0;
  double acquire_wait_ms_ = 0.0;  // cumulative blocked-in-REQ time
  bool holding_ = false;
  bool hint_enabled_ = true;
  double grant_time_ = 0.0;
  double ewma_used_ = 0.0;

 public:
  long long failures_ = 0;
  long long idle_releases_ = 0;
};

}  // namespace ks
