// gpu-schd core: per-GPU sliding-window token scheduler (policy only,
// no I/O — unit-testable; the poll loop lives in gpu_schd.cpp).
//
// Role model: the reference's gem-schd (one instance per physical GPU,
// CLI flags -q 300 -m 20 -w 10000, SURVEY.md §2.2 and
// docker/kubeshare-gemini-scheduler/launcher.py:27-31,77-80). Internals
// are re-derived: stride-based work-conserving time slicing.
//
// Model
// -----
// At most ONE token is outstanding per GPU. A token is a wall-time lease
// of quota_ms during which the holder may submit GPU work freely; the
// holder drains its streams at expiry and reports measured use (RET).
// Co-located pods therefore run in alternating near-exclusive windows —
// on MI355X there is no enforced SR-IOV partitioning inside one GPU, so
// time-multiplexing the 256 CUs at ~300 ms granularity is both the
// simplest and the highest-throughput isolation mechanism (each slice
// runs with full HBM bandwidth and all CUs; no cache thrash from
// co-running kernels).
//
// Fairness policy (per grant decision) — stride/virtual-time:
//   usage(pod)  = sum of charged ms inside the sliding window w
//   over(pod)   = usage >= limit*w            -> ineligible (hard cap)
//   v(pod)      = usage / weight              -> LOWEST virtual time wins
//                 (weight = gpu_request; opportunistic request-0 pods
//                 get a tiny epsilon weight: they run only ahead of
//                 nobody, but are never starved forever — the window
//                 prunes their v back down)
//   tie         = FIFO
//   quota       = clamp(min(hint-sized base, limit*w - usage), min_q, base_q)
//
// Why stride and not max-absolute-deficit: with deficit ordering a
// big-request pod monopolizes the GPU for O(window x request) at every
// window turnover before a small-request pod's deficit can compete —
// measured as recurring 0.6-2.3 s p99 latency spikes for a co-located
// serving pod (tools/serve_probe.py, round 2). Virtual time interleaves
// grants in proportion to requests at every scale: continuously hungry
// pods converge to usage_i/request_i equal, i.e. exactly the requested
// split, and a pod returning from idle has low v (immediate service,
// bounded by its own share). Work conservation is unchanged: whoever
// is eligible and hungriest-by-v runs; a solo pod bursts to limit.
#pragma once

#include <algorithm>
#include <deque>
#include <map>
#include <set>
#include <string>
#include <vector>

#include "../common/protocol.hpp"

namespace ks {

struct PodAccount {
  double request = 0.0;
  double limit = 1.0;
  long long memory = 0;       // byte cap (0 = default)
  double lease_ms = 0.0;      // latency class: lease override (0 = off)
  std::string group;          // gang group ("" = none)
  bool in_config = false;     // listed in the per-UUID config file?
  // (t_charge_end_ms, used_ms) entries inside the window
  std::deque<std::pair<double, double>> charges;
  double usage_cache = 0.0;
  // lifetime stats
  double total_used_ms = 0.0;
  long long grants = 0;
  long long mem_bytes = 0;    // last reported footprint

  void prune(double now, double window_ms) {
    while (!charges.empty() && charges.front().first < now - window_ms) {
      usage_cache -= charges.front().second;
      charges.pop_front();
    }
    if (charges.empty()) usage_cache = 0.0;
  }
  void charge(double now, double used_ms) {
    charges.emplace_back(now, used_ms);
    usage_cache += used_ms;
    total_used_ms += used_ms;
  }
};

struct Waiter {
  std::string pod;
  long long cookie;   // opaque per-request id assigned by the I/O layer
  double hint_ms;
  double enq_ms;
};

struct Grant {
  std::string pod;
  long long cookie = -1;
  double quota_ms = 0.0;
};

class TokenScheduler {
 public:
  TokenScheduler(double base_quota_ms, double min_quota_ms, double window_ms)
      : base_q_(base_quota_ms), min_q_(min_quota_ms), window_(window_ms) {}

  void set_config(const std::vector<PodQuota>& quotas,
                  double now = -1.0) {
    for (auto& kv : pods_) kv.second.in_config = false;
    for (const auto& q : quotas) {
      PodAccount& a = pods_[q.pod];
      a.request = q.request;
      a.limit = q.limit;
      a.memory = q.memory;
      a.lease_ms = q.lease_ms;
      a.group = q.group;
      a.in_config = true;
    }
    // prune accounts of departed pods once their window usage has
    // aged out (no holder, no charges) — a node churning pods for
    // months must not grow pods_ unboundedly (measured: ~840 B per
    // pod lifetime before this)
    if (now < 0.0) return;  // virtual-clock callers skip pruning
    for (auto it = pods_.begin(); it != pods_.end();) {
      PodAccount& a = it->second;
      a.prune(now, window_);
      if (!a.in_config && a.charges.empty() &&
          holders_.find(it->first) == holders_.end()) {
        it = pods_.erase(it);
      } else {
        ++it;
      }
    }
  }

  PodAccount& account(const std::string& pod) {
    auto it = pods_.find(pod);
    if (it != pods_.end()) return it->second;
    // Unknown pod (config file may lag pod start — the reference's
    // Prometheus round-trip had the same gap): run it opportunistically
    // with no guarantee and full burst until the config lands.
    PodAccount a;
    a.request = 0.0;
    a.limit = 1.0;
    pods_[pod] = a;
    return pods_[pod];
  }

  // A pod asked for a token.
  void request(const std::string& pod, long long cookie, double hint_ms,
               double now) {
    waiters_.push_back(Waiter{pod, cookie, hint_ms, now});
  }

  // A holder returned its token. `used_ms` is the client-measured GPU
  // time; it is clamped server-side to [0, wall + slack] so a lying
  // client cannot under-report what it actually occupied. When the
  // server-side busy sampler is feeding add_busy(), the SAMPLED busy
  // time is charged instead — a CPU-bound phase inside a lease then
  // costs only the GPU time it actually used, and the client's report
  // becomes advisory (quota-error measurement stays honest even
  // against a hostile hook).
  void release(const std::string& pod, double used_ms, double now) {
    auto it = holders_.find(pod);
    if (it != holders_.end()) {
      double wall = now - it->second.grant_ms;
      double charged;
      if (sampler_active_) {
        charged = std::min(it->second.sampled_ms, wall + 50.0);
      } else {
        charged = std::max(0.0, std::min(used_ms, wall + 50.0));
        // a GPU-bound holder's wall time is the true exclusive occupancy
        charged = std::max(charged, std::min(wall, it->second.quota));
      }
      account(pod).charge(now, charged);
      holders_.erase(it);
      if (holders_.empty()) holder_group_.clear();
    } else {
      // late RET after a revocation: already charged, ignore amount
    }
  }

  // Server-side GPU-busy feed (gpu-schd samples rocm_smi's
  // gpu_busy_percent between poll-loop ticks). Attributed to the
  // current holder(s); co-granted gang members split the interval
  // evenly (they share one GPU by construction). Unattributed busy
  // (no holder: RCCL-exempt kernels, ungated processes) is tracked for
  // stats only.
  void add_busy(double busy_ms, double now) {
    (void)now;
    sampler_active_ = true;
    if (holders_.empty()) {
      other_busy_ms_ += busy_ms;
      return;
    }
    double share = busy_ms / (double)holders_.size();
    for (auto& kv : holders_) kv.second.sampled_ms += share;
  }

  bool sampler_active() const { return sampler_active_; }

  // Holder liveness: revoke tokens whose holders exceeded their lease
  // (died or hung; the connection may still be open).
  bool check_revoke(double now, double* revoked_at = nullptr) {
    bool any = false;
    for (auto it = holders_.begin(); it != holders_.end();) {
      double deadline = it->second.grant_ms + it->second.quota * 3.0 + 2000.0;
      if (now >= deadline) {
        account(it->first).charge(
            now, sampler_active_ ? it->second.sampled_ms
                                 : now - it->second.grant_ms);
        last_revoked_ = it->first;
        revokes_++;
        it = holders_.erase(it);
        any = true;
      } else {
        ++it;
      }
    }
    if (holders_.empty()) holder_group_.clear();
    if (any && revoked_at) *revoked_at = now;
    return any;
  }

  // Drop the EXACT waiters a dead connection owns. (Cookies are global
  // and interleave across connections — a range here once dropped other
  // connections' pending REQs, stranding their clients.)
  void drop_waiters(const std::set<long long>& cookies) {
    waiters_.erase(std::remove_if(waiters_.begin(), waiters_.end(),
                                  [&](const Waiter& w) {
                                    return cookies.count(w.cookie) > 0;
                                  }),
                   waiters_.end());
  }

  // If a holder vanished without RET (connection closed), charge wall
  // (or sampled busy when the sampler runs).
  void force_release(const std::string& pod, double now) {
    auto it = holders_.find(pod);
    if (it != holders_.end()) {
      account(pod).charge(
          now, sampler_active_ ? it->second.sampled_ms
                               : now - it->second.grant_ms);
      holders_.erase(it);
      if (holders_.empty()) holder_group_.clear();
    }
  }

  // Record which connection's REQ produced a live grant, so a DYING
  // connection only frees grants it owns. (Round-1 advisor finding:
  // keying the cleanup by pod name could free a token a still-live
  // connection of the same pod was holding — two concurrent holders.)
  void set_holder_owner(const std::string& pod, long long owner) {
    auto it = holders_.find(pod);
    if (it != holders_.end()) it->second.owner = owner;
  }

  // A connection died: free exactly the grants it owns.
  void force_release_owned(long long owner, double now) {
    for (auto it = holders_.begin(); it != holders_.end();) {
      if (it->second.owner == owner) {
        account(it->first).charge(
            now, sampler_active_ ? it->second.sampled_ms
                                 : now - it->second.grant_ms);
        it = holders_.erase(it);
      } else {
        ++it;
      }
    }
    if (holders_.empty()) holder_group_.clear();
  }

  // Pick the next holder. Returns true and fills `out` when a token was
  // granted; when false and `next_try_ms` > 0, every waiter is over its
  // hard cap and the caller should re-run at now+next_try_ms. Called in
  // a loop by the I/O layer, so GANG CO-GRANTING falls out naturally:
  // while a gang's members hold the token, further members of the SAME
  // group are granted immediately (a DDP collective in one rank must
  // never spin on a token-starved peer — docs/ROADMAP.md), and everyone
  // else waits until the whole gang has drained.
  bool schedule(double now, Grant* out, double* next_try_ms) {
    *next_try_ms = 0.0;
    if (waiters_.empty()) return false;
    const bool gang_active = !holders_.empty() && !holder_group_.empty();
    if (!holders_.empty() && !gang_active) return false;

    double best_key = -1e300;
    int best_i = -1;
    double soonest = 1e300;
    for (int i = 0; i < (int)waiters_.size(); i++) {
      PodAccount& a = account(waiters_[i].pod);
      if (!holders_.empty()) {
        // only co-grant same-gang members (never the same pod twice)
        if (a.group.empty() || a.group != holder_group_ ||
            holders_.count(waiters_[i].pod))
          continue;
      }
      a.prune(now, window_);
      // limit >= 1.0 means "may use the whole GPU": no hard cap. (A
      // busy solo pod's window usage approaches window_ by definition;
      // clamping it to ever-smaller leases would thrash on drains.)
      bool uncapped = a.limit >= 0.999;
      double cap = a.limit * window_;
      if (!uncapped && a.usage_cache >= cap - 1e-9) {
        // over hard cap: eligible again when the oldest charge ages out
        if (!a.charges.empty())
          soonest = std::min(soonest,
                             a.charges.front().first + window_ - now);
        continue;
      }
      // stride scheduling: lowest virtual time (usage/weight) wins
      double weight = a.request > 0.0 ? a.request : 0.01;
      double key = -(a.usage_cache / weight);
      // FIFO tiebreak: earlier enqueue wins on exact ties
      key -= (waiters_[i].enq_ms - now) * 1e-12;
      if (key > best_key) {
        best_key = key;
        best_i = i;
      }
    }
    if (best_i < 0) {
      *next_try_ms = std::max(1.0, std::min(soonest, window_));
      return false;
    }

    Waiter w = waiters_[best_i];
    waiters_.erase(waiters_.begin() + best_i);
    PodAccount& a = account(w.pod);
    double room = a.limit >= 0.999 ? base_q_
                                   : a.limit * window_ - a.usage_cache;
    // hint_ms = the client's EWMA of busy time per lease. A bursty pod
    // that keeps releasing early gets right-sized (smaller) leases —
    // finer-grained alternation, less over-grant — while the x1.5
    // headroom lets a ramping pod grow back to base_q within a few
    // leases. Always clamped to [min_q, base_q].
    double desired = base_q_;
    if (a.lease_ms > 0.0)
      desired = a.lease_ms;  // explicit latency class beats the hint
    else if (w.hint_ms > 0.0)
      desired = std::min(base_q_, std::max(min_q_, w.hint_ms * 1.5));
    desired = std::min(desired, base_q_);
    double quota = std::min(desired, room);
    quota = std::max(quota, min_q_);
    holders_[w.pod] = Holder{now, quota};
    holder_group_ = a.group;
    a.grants++;
    out->pod = w.pod;
    out->cookie = w.cookie;
    out->quota_ms = quota;
    return true;
  }

  size_t n_holders() const { return holders_.size(); }
  size_t n_waiters() const { return waiters_.size(); }

  // One-line JSON stats: per-pod window usage, share of busy time,
  // quota error vs request. Used by bench.py and the GPU tests as the
  // *server-side* quota-enforcement measurement.
  std::string stats_json(double now) {
    double total_busy = 0.0;
    for (auto& kv : pods_) {
      kv.second.prune(now, window_);
      total_busy += kv.second.usage_cache;
    }
    std::string holders;
    for (auto& kv : holders_) {
      if (!holders.empty()) holders += ",";
      holders += "\"" + jesc(kv.first) + "\":" +
                 fmt(now - kv.second.grant_ms);
    }
    std::string s = "{\"window_ms\":" + fmt(window_) +
                    ",\"busy_ms\":" + fmt(total_busy) +
                    ",\"sampler\":" + (sampler_active_ ? "true" : "false") +
                    ",\"other_busy_ms\":" + fmt(other_busy_ms_) +
                    ",\"revokes\":" + std::to_string(revokes_) +
                    ",\"last_revoked\":\"" + jesc(last_revoked_) + "\"" +
                    ",\"holders\":{" + holders + "}" +
                    ",\"waiters\":" + std::to_string(waiters_.size()) +
                    ",\"pods\":{";
    bool first = true;
    for (auto& kv : pods_) {
      const PodAccount& a = kv.second;
      if (!first) s += ",";
      first = false;
      double share = total_busy > 0 ? a.usage_cache / total_busy : 0.0;
      double frac = a.usage_cache / window_;
      s += "\"" + jesc(kv.first) + "\":{\"request\":" + fmt(a.request) +
           ",\"limit\":" + fmt(a.limit) + ",\"usage_ms\":" +
           fmt(a.usage_cache) + ",\"window_frac\":" + fmt(frac) +
           ",\"busy_share\":" + fmt(share) + ",\"grants\":" +
           std::to_string(a.grants) + ",\"total_used_ms\":" +
           fmt(a.total_used_ms) + ",\"mem_bytes\":" +
           std::to_string(a.mem_bytes) + "}";
    }
    s += "}}";
    return s;
  }

 private:
  static std::string fmt(double v) {
    char b[32];
    snprintf(b, sizeof(b), "%.4f", v);
    return b;
  }

  // Pod names are k8s-validated in the cluster path, but local-mode
  // REQs (hook loopback, no pod-mgr stamping) carry arbitrary bytes —
  // they must not corrupt the STATS JSON reply.
  static std::string jesc(const std::string& s) {
    std::string out;
    out.reserve(s.size());
    for (char c : s) {
      if (c == '"' || c == '\\') out += '\\';
      if ((unsigned char)c < 0x20) { out += '?'; continue; }
      out += c;
    }
    return out;
  }

  struct Holder {
    double grant_ms = 0.0;
    double quota = 0.0;
    long long owner = -1;     // connection id whose REQ was granted
    double sampled_ms = 0.0;  // server-sampled GPU-busy inside the lease
  };

  double base_q_, min_q_, window_;
  std::map<std::string, PodAccount> pods_;
  std::vector<Waiter> waiters_;
  std::map<std::string, Holder> holders_;
  std::string holder_group_;  // group of the current holders ("": none)
  bool sampler_active_ = false;
  double other_busy_ms_ = 0.0;  // sampled busy with no holder (stats)
  long long revokes_ = 0;       // liveness revocations (lost RETs)
  std::string last_revoked_;
};

}  // namespace ks
