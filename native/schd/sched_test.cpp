// Deterministic unit tests for the TokenScheduler policy (virtual
// clock — every TokenScheduler entry point takes `now`, so fairness,
// caps, decay, revocation and gang co-granting are tested without a
// single sleep). Run by tests/test_native_daemons.py::test_sched_unit.
#include <cassert>
#include <cstdio>

#include "token_sched.hpp"

using namespace ks;

static int g_checks = 0;
#define CHECK(cond)                                                       \
  do {                                                                    \
    if (!(cond)) {                                                        \
      fprintf(stderr, "CHECK failed at %s:%d: %s\n", __FILE__, __LINE__,  \
              #cond);                                                     \
      return 1;                                                           \
    }                                                                     \
    g_checks++;                                                           \
  } while (0)

static PodQuota quota(const char* pod, double limit, double request,
                      const char* group = "") {
  PodQuota q;
  q.pod = pod;
  q.limit = limit;
  q.request = request;
  q.group = group;
  return q;
}

int main() {
  // ---- 1. solo pod: work-conserving regrant, usage accounting
  {
    TokenScheduler s(300, 20, 10000);
    s.set_config({quota("a", 1.0, 0.5)});
    Grant g;
    double retry;
    s.request("a", 1, 0, 0.0);
    CHECK(s.schedule(0.0, &g, &retry));
    CHECK(g.pod == "a" && g.quota_ms == 300);
    CHECK(!s.schedule(0.0, &g, &retry));  // one holder, no co-grant (no gang)
    s.release("a", 300, 300.0);
    CHECK(s.n_holders() == 0);
    s.request("a", 2, 0, 300.0);
    CHECK(s.schedule(300.0, &g, &retry));  // immediate regrant
    s.release("a", 300, 600.0);
  }

  // ---- 2. deficit fairness: the under-served pod wins
  {
    TokenScheduler s(100, 20, 10000);
    s.set_config({quota("a", 1.0, 0.7), quota("b", 1.0, 0.3)});
    double t = 0;
    double used_a = 0, used_b = 0;
    for (int i = 0; i < 100; i++) {
      s.request("a", i * 2, 0, t);
      s.request("b", i * 2 + 1, 0, t);
      Grant g;
      double retry;
      CHECK(s.schedule(t, &g, &retry));
      t += g.quota_ms;
      (g.pod == "a" ? used_a : used_b) += g.quota_ms;
      s.release(g.pod, g.quota_ms, t);
      // drop the loser's stale waiter for the next round
      s.drop_waiters({i * 2, i * 2 + 1});
    }
    double share_a = used_a / (used_a + used_b);
    CHECK(share_a > 0.65 && share_a < 0.75);
  }

  // ---- 3. hard cap + window decay
  {
    TokenScheduler s(100, 20, 1000);  // 1 s window
    s.set_config({quota("c", 0.4, 0.2)});
    Grant g;
    double retry;
    double t = 0;
    double granted = 0;
    // burn up to the cap: 0.4 * 1000 = 400 ms of room
    for (int i = 0; i < 10; i++) {
      s.request("c", 100 + i, 0, t);
      if (!s.schedule(t, &g, &retry)) break;
      granted += g.quota_ms;
      t += g.quota_ms;
      s.release("c", g.quota_ms, t);
    }
    CHECK(granted >= 380 && granted <= 420);
    // at the cap: refused with a retry hint
    s.request("c", 200, 0, t);
    CHECK(!s.schedule(t, &g, &retry));
    CHECK(retry > 0);
    // after the window decays, eligible again
    CHECK(s.schedule(t + 1100.0, &g, &retry));
    CHECK(g.pod == "c");
  }

  // ---- 4. min-quota clamp near the cap
  {
    TokenScheduler s(300, 20, 1000);
    s.set_config({quota("d", 0.31, 0.31)});
    Grant g;
    double retry;
    s.request("d", 1, 0, 0.0);
    CHECK(s.schedule(0.0, &g, &retry));
    CHECK(g.quota_ms == 300);  // room 310 -> full base quota
    s.release("d", 300, 300.0);
    s.request("d", 2, 0, 300.0);
    CHECK(s.schedule(300.0, &g, &retry));
    CHECK(g.quota_ms == 20);  // room 10 -> clamped up to min quota
  }

  // ---- 5. liveness revocation frees a wedged holder
  {
    TokenScheduler s(100, 20, 10000);
    s.set_config({quota("a", 1.0, 0.5), quota("b", 1.0, 0.5)});
    Grant g;
    double retry;
    s.request("a", 1, 0, 0.0);
    CHECK(s.schedule(0.0, &g, &retry));
    s.request("b", 2, 0, 10.0);
    CHECK(!s.schedule(10.0, &g, &retry));
    // deadline = quota*3 + 2000 = 2300
    CHECK(!s.check_revoke(2200.0));
    CHECK(s.check_revoke(2400.0));
    CHECK(s.schedule(2400.0, &g, &retry));
    CHECK(g.pod == "b");
  }

  // ---- 6. gang co-granting
  {
    TokenScheduler s(200, 20, 10000);
    s.set_config({quota("g0", 1.0, 0.3, "ddp"), quota("g1", 1.0, 0.3, "ddp"),
                  quota("solo", 1.0, 0.4)});
    Grant g;
    double retry;
    s.request("g0", 1, 0, 0.0);
    s.request("solo", 2, 0, 0.0);
    s.request("g1", 3, 0, 1.0);
    CHECK(s.schedule(1.0, &g, &retry));
    std::string first = g.pod;
    CHECK(first == "g0" || first == "solo");  // deficit order
    if (first == "solo") {
      // drain solo, then the gang pair must co-hold
      s.release("solo", 200, 201.0);
      CHECK(s.schedule(201.0, &g, &retry));
      CHECK(g.pod == "g0" || g.pod == "g1");
    }
    // whoever of the gang holds, the partner is co-granted...
    CHECK(s.schedule(202.0, &g, &retry));
    CHECK(s.n_holders() == 2);
    // ...and nobody else is until the WHOLE gang drains
    s.request("solo", 4, 0, 203.0);
    CHECK(!s.schedule(203.0, &g, &retry));
    s.release("g0", 100, 300.0);
    CHECK(!s.schedule(300.0, &g, &retry));  // g1 still holds
    s.release("g1", 100, 310.0);
    CHECK(s.schedule(310.0, &g, &retry));
    CHECK(g.pod == "solo");
  }

  // ---- 7. exact-cookie waiter cancellation
  {
    TokenScheduler s(100, 20, 10000);
    s.set_config({quota("a", 1.0, 0.5), quota("b", 1.0, 0.5)});
    Grant g;
    double retry;
    s.request("a", 10, 0, 0.0);
    CHECK(s.schedule(0.0, &g, &retry));   // a holds
    s.request("a", 11, 0, 1.0);           // a's next REQ
    s.request("b", 12, 0, 2.0);           // b waits
    s.drop_waiters({11});                 // a's CONNECTION died
    s.force_release("a", 3.0);
    CHECK(s.schedule(3.0, &g, &retry));
    CHECK(g.pod == "b" && g.cookie == 12);  // b unharmed
  }

  // ---- 8. uncapped limit >= 1.0 never degrades the quota
  {
    TokenScheduler s(300, 20, 1000);  // tiny window vs base quota
    s.set_config({quota("e", 1.0, 0.5)});
    Grant g;
    double retry;
    double t = 0;
    for (int i = 0; i < 20; i++) {
      s.request("e", i, 0, t);
      CHECK(s.schedule(t, &g, &retry));
      CHECK(g.quota_ms == 300);  // never clamped toward min quota
      t += g.quota_ms;
      s.release("e", g.quota_ms, t);
    }
  }

  // ---- 9. late RET after revocation is ignored (no double charge)
  {
    TokenScheduler s(100, 20, 10000);
    s.set_config({quota("a", 1.0, 0.5)});
    Grant g;
    double retry;
    s.request("a", 1, 0, 0.0);
    CHECK(s.schedule(0.0, &g, &retry));
    CHECK(s.check_revoke(5000.0));  // charged wall = 5000
    double before = s.account("a").total_used_ms;
    s.release("a", 400.0, 5100.0);  // late RET from the revoked holder
    CHECK(s.account("a").total_used_ms == before);
  }

  // ---- 10. unknown pod runs opportunistically, then the config lands
  {
    TokenScheduler s(100, 20, 10000);
    Grant g;
    double retry;
    s.request("ghost", 1, 0, 0.0);
    CHECK(s.schedule(0.0, &g, &retry));  // request 0 / limit 1 defaults
    CHECK(g.pod == "ghost");
    s.release("ghost", 100, 100.0);
    s.set_config({quota("ghost", 0.5, 0.5)});
    CHECK(s.account("ghost").request == 0.5);
    CHECK(s.account("ghost").in_config);
    // earlier usage survives the reload
    CHECK(s.account("ghost").total_used_ms == 100.0);
  }

  // ---- 11. a lying client cannot under-report its occupancy
  {
    TokenScheduler s(100, 20, 10000);
    s.set_config({quota("liar", 1.0, 0.5)});
    Grant g;
    double retry;
    s.request("liar", 1, 0, 0.0);
    CHECK(s.schedule(0.0, &g, &retry));
    // held 100 ms of wall but claims 1 ms: charged >= min(wall, quota)
    s.release("liar", 1.0, 100.0);
    CHECK(s.account("liar").total_used_ms >= 100.0);
    // ...nor wildly over-report (clamped to wall + slack)
    s.request("liar", 2, 0, 100.0);
    CHECK(s.schedule(100.0, &g, &retry));
    s.release("liar", 99999.0, 150.0);
    CHECK(s.account("liar").total_used_ms <= 100.0 + 50.0 + 51.0);
  }

  // ---- 12. stats JSON carries the accounting fields
  {
    TokenScheduler s(100, 20, 10000);
    s.set_config({quota("j", 0.8, 0.6)});
    Grant g;
    double retry;
    s.request("j", 1, 0, 0.0);
    CHECK(s.schedule(0.0, &g, &retry));
    s.release("j", 100, 100.0);
    std::string js = s.stats_json(100.0);
    CHECK(js.find("\"j\"") != std::string::npos);
    CHECK(js.find("\"request\":0.6000") != std::string::npos);
    CHECK(js.find("\"grants\":1") != std::string::npos);
  }

  // ---- 13. grant-owner tracking: a dying sibling connection must not
  // free a grant another connection holds (round-1 advisor finding)
  {
    TokenScheduler s(100, 20, 10000);
    s.set_config({quota("a", 1.0, 0.5)});
    Grant g;
    double retry;
    s.request("a", 1, 0, 0.0);
    CHECK(s.schedule(0.0, &g, &retry));
    s.set_holder_owner("a", 7);     // conn 7 produced the grant
    s.force_release_owned(9, 10.0);  // conn 9 (same pod) dies
    CHECK(s.n_holders() == 1);       // grant untouched
    s.force_release_owned(7, 50.0);  // the owning conn dies
    CHECK(s.n_holders() == 0);
    CHECK(s.account("a").total_used_ms == 50.0);  // charged wall
  }

  // ---- 14. hint-sized quotas: EWMA hint right-sizes the lease
  {
    TokenScheduler s(300, 20, 10000);
    s.set_config({quota("a", 1.0, 0.5)});
    Grant g;
    double retry;
    s.request("a", 1, 50.0, 0.0);   // hint 50 -> quota 75 (x1.5)
    CHECK(s.schedule(0.0, &g, &retry));
    CHECK(g.quota_ms == 75.0);
    s.release("a", 75, 75.0);
    s.request("a", 2, 3.0, 75.0);   // tiny hint clamps to min_q
    CHECK(s.schedule(75.0, &g, &retry));
    CHECK(g.quota_ms == 20.0);
    s.release("a", 20, 95.0);
    s.request("a", 3, 0.0, 95.0);   // no hint -> base quota
    CHECK(s.schedule(95.0, &g, &retry));
    CHECK(g.quota_ms == 300.0);
    s.release("a", 300, 395.0);
    s.request("a", 4, 5000.0, 395.0);  // huge hint clamps to base_q
    CHECK(s.schedule(395.0, &g, &retry));
    CHECK(g.quota_ms == 300.0);
  }

  // ---- 15. server-side busy sampling: charge sampled GPU time, not
  // wall — a CPU-bound phase inside the lease costs nothing
  {
    TokenScheduler s(300, 20, 10000);
    s.set_config({quota("a", 1.0, 0.5)});
    Grant g;
    double retry;
    s.request("a", 1, 0, 0.0);
    CHECK(s.schedule(0.0, &g, &retry));
    // 300 ms of wall, but the sampler saw only 90 ms of GPU activity
    for (int i = 0; i < 30; i++) s.add_busy(3.0, i * 10.0);
    s.release("a", 300.0, 300.0);  // client reports wall; ignored
    CHECK(s.account("a").total_used_ms == 90.0);
    // unattributed busy (no holder) is not charged to anyone
    double before = s.account("a").total_used_ms;
    s.add_busy(50.0, 400.0);
    CHECK(s.account("a").total_used_ms == before);
    CHECK(s.stats_json(400.0).find("\"sampler\":true") != std::string::npos);
  }

  // ---- 16. sampled busy splits evenly across gang co-holders
  {
    TokenScheduler s(300, 20, 10000);
    s.set_config({quota("g1", 1.0, 0.5, "gang"), quota("g2", 1.0, 0.5,
                                                       "gang")});
    Grant g;
    double retry;
    s.request("g1", 1, 0, 0.0);
    s.request("g2", 2, 0, 0.0);
    CHECK(s.schedule(0.0, &g, &retry));
    CHECK(s.schedule(0.0, &g, &retry));  // co-granted (same gang)
    CHECK(s.n_holders() == 2);
    s.add_busy(100.0, 50.0);
    s.release("g1", 300.0, 300.0);
    s.release("g2", 300.0, 300.0);
    CHECK(s.account("g1").total_used_ms == 50.0);
    CHECK(s.account("g2").total_used_ms == 50.0);
  }

  // ---- 17. no window-turnover starvation of a small-request pod
  // (round-2 regression: with absolute-deficit ordering a 0.7-request
  // trainer monopolized O(window x request) at every window turnover,
  // measured as 0.6-2.3 s serving p99 spikes; stride ordering bounds
  // the sparse pod's wait to ~one trainer lease)
  {
    TokenScheduler s(50, 10, 4000);
    s.set_config({quota("train", 1.0, 0.7), quota("serve", 1.0, 0.3)});
    double t = 0.0;
    double next_serve_req = 0.0;     // serving request every 60 ms
    double serve_req_at = -1.0;      // outstanding serve REQ time
    double worst_wait = 0.0;
    long long ck = 1;
    for (int iter = 0; iter < 4000; iter++) {
      if (serve_req_at < 0.0 && t >= next_serve_req) {
        s.request("serve", ck++, 5.0, t);
        serve_req_at = t;
      }
      s.request("train", ck++, 0.0, t);
      Grant g;
      double retry;
      if (!s.schedule(t, &g, &retry)) {
        t += 1.0;
        continue;
      }
      if (g.pod == "serve") {
        worst_wait = std::max(worst_wait, t - serve_req_at);
        double used = 5.0;           // short inference burst
        t += used;
        s.release("serve", used, t);
        serve_req_at = -1.0;
        next_serve_req = t + 60.0;
      } else {
        t += g.quota_ms;             // trainer saturates its lease
        s.release("train", g.quota_ms, t);
      }
    }
    // deficit ordering yields worst waits of ~1600 ms here; stride
    // keeps it near one trainer lease
    CHECK(worst_wait <= 120.0);
    // and the trainer still gets the bulk of the GPU
    PodAccount& tr = s.account("train");
    PodAccount& sv = s.account("serve");
    CHECK(tr.total_used_ms > 4.0 * sv.total_used_ms);
  }

  // ---- 18. opportunistic (request-0) pod vs a saturated guaranteed
  // pod: stride gives it a bounded anti-starvation trickle (~weight
  // epsilon/request ratio), full burst only when the GPU is otherwise
  // idle
  {
    TokenScheduler s(50, 10, 4000);
    s.set_config({quota("guar", 1.0, 0.7), quota("opp", 1.0, 0.0)});
    double t = 0.0;
    long long ck = 1;
    for (int iter = 0; iter < 2000; iter++) {
      s.request("guar", ck++, 0.0, t);
      s.request("opp", ck++, 0.0, t);
      Grant g;
      double retry;
      if (!s.schedule(t, &g, &retry)) {
        t += 1.0;
        continue;
      }
      t += g.quota_ms;
      s.release(g.pod, g.quota_ms, t);
      // drop stale waiters so the queue doesn't grow across rounds
      s.drop_waiters({ck - 1, ck - 2});
    }
    double u_g = s.account("guar").total_used_ms;
    double u_o = s.account("opp").total_used_ms;
    CHECK(u_g > 0 && u_o > 0);               // not starved entirely
    CHECK(u_o / (u_g + u_o) < 0.26);         // min-quota-floored trickle
    // alone on the GPU, the opportunistic pod bursts freely
    s.request("opp", ck++, 0.0, t);
    Grant g;
    double retry;
    CHECK(s.schedule(t, &g, &retry));
    CHECK(g.pod == "opp");
  }

  // ---- 19. latency class: per-pod lease override (config q=<ms>)
  {
    TokenScheduler s(300, 10, 10000);
    PodQuota q1 = quota("svc", 1.0, 0.3);
    q1.lease_ms = 25.0;
    s.set_config({q1, quota("train", 1.0, 0.7)});
    Grant g;
    double retry;
    s.request("svc", 1, 200.0, 0.0);  // large hint must NOT win
    CHECK(s.schedule(0.0, &g, &retry));
    CHECK(g.pod == "svc" && g.quota_ms == 25.0);
    s.release("svc", 25, 25.0);
    s.request("train", 2, 0.0, 25.0);  // no override -> base quota
    CHECK(s.schedule(25.0, &g, &retry));
    CHECK(g.quota_ms == 300.0);
    s.release("train", 300, 325.0);
    // override above base clamps to base; below min clamps to min
    PodQuota q2 = quota("hi", 1.0, 0.2);
    q2.lease_ms = 900.0;
    PodQuota q3 = quota("lo", 1.0, 0.2);
    q3.lease_ms = 1.0;
    s.set_config({q2, q3});
    s.request("hi", 3, 0.0, 325.0);
    CHECK(s.schedule(325.0, &g, &retry));
    CHECK(g.quota_ms == 300.0);
    s.release("hi", 10, 335.0);
    s.request("lo", 4, 0.0, 335.0);
    CHECK(s.schedule(335.0, &g, &retry));
    CHECK(g.quota_ms == 10.0);  // min_q floor
  }

  // ---- 20. config-line parser: optional trailing fields in any order
  {
    FILE* f = tmpfile();
    fputs("4\n"
          "ns/a 1.0 0.5 0\n"
          "ns/b 1.0 0.5 123 gangx\n"
          "ns/c 1.0 0.5 0 q=25\n"
          "ns/d 0.9 0.4 7 q=15 gangy\n",  // swapped order also legal
          f);
    rewind(f);
    std::vector<PodQuota> out;
    CHECK(parse_gpu_config(f, out));
    fclose(f);
    CHECK(out.size() == 4);
    CHECK(out[0].group.empty() && out[0].lease_ms == 0.0);
    CHECK(out[1].group == "gangx" && out[1].lease_ms == 0.0);
    CHECK(out[2].group.empty() && out[2].lease_ms == 25.0);
    CHECK(out[3].group == "gangy" && out[3].lease_ms == 15.0);
    CHECK(out[3].memory == 7 && out[3].request == 0.4);
  }

  // ---- 21. departed-pod account pruning (live-clock set_config)
  {
    TokenScheduler s(50, 10, 1000);
    s.set_config({quota("old", 1.0, 0.5)}, 0.0);
    Grant g;
    double retry;
    s.request("old", 1, 0, 0.0);
    CHECK(s.schedule(0.0, &g, &retry));
    s.release("old", 50, 50.0);
    // pod leaves the config while its usage is still in the window:
    // account is retained (anti-gaming: usage must keep counting)
    s.set_config({quota("new", 1.0, 0.5)}, 100.0);
    CHECK(s.stats_json(100.0).find("\"old\"") != std::string::npos);
    // after the window ages the charge out, the account is pruned
    s.set_config({quota("new", 1.0, 0.5)}, 2000.0);
    CHECK(s.stats_json(2000.0).find("\"old\"") == std::string::npos);
    // a pod still HOLDING is never pruned even out-of-config
    s.request("new", 2, 0, 2000.0);
    CHECK(s.schedule(2000.0, &g, &retry));
    s.set_config({}, 9999999.0);
    CHECK(s.n_holders() == 1);
    CHECK(s.stats_json(9999999.0).find("\"new\"") != std::string::npos);
  }

  // ---- 22. hostile pod names never corrupt the STATS JSON
  {
    TokenScheduler s(50, 10, 1000);
    Grant g;
    double retry;
    // local-mode REQ: name with quote, backslash and a control byte
    s.request("ns/evil\"pod\\x\n", 1, 0, 0.0);
    CHECK(s.schedule(0.0, &g, &retry));
    std::string j = s.stats_json(10.0);
    // every unescaped quote must be a key/value delimiter: scan for
    // balanced structure — no raw '"' preceded by a name byte
    CHECK(j.find("evil\\\"pod\\\\x?") != std::string::npos);
    CHECK(j.find('\n') == std::string::npos);
    // revocation path escapes too
    CHECK(s.check_revoke(100000.0));
    CHECK(s.stats_json(100000.0).find("\"last_revoked\":\"ns/evil\\\"") !=
          std::string::npos);
  }

  printf("sched_test OK (%d checks)\n", g_checks);
  return 0;
}
