// fairness_sweep — virtual-clock sweep of the stride policy across
// request/limit mixes (results: profiles/stride_fairness_sweep.txt).
//   make -C native fairness_sweep && native/fairness_sweep
#include <cstdio>

#include "token_sched.hpp"

using namespace ks;

int main() {
  printf("# reqA reqB limA limB -> shareA shareB (util)\n");
  double mixes[][4] = {{0.9, 0.1, 1, 1}, {0.7, 0.3, 1, 1},
                       {0.5, 0.5, 1, 1}, {0.6, 0.2, 1, 1},
                       {0.5, 0.25, 0.5, 0.25}, {0.3, 0.3, 0.4, 0.4},
                       {0.8, 0.0, 1, 1}, {0.34, 0.33, 1, 1}};
  for (auto& m : mixes) {
    TokenScheduler s(300, 20, 10000);
    PodQuota a, b;
    a.pod = "A"; a.request = m[0]; a.limit = m[2];
    b.pod = "B"; b.request = m[1]; b.limit = m[3];
    s.set_config({a, b});
    double t = 0;
    long long ck = 1;
    double used[2] = {0, 0};
    while (t < 120000) {  // 2 virtual minutes
      s.request("A", ck++, 0, t);
      s.request("B", ck++, 0, t);
      Grant g;
      double retry;
      if (!s.schedule(t, &g, &retry)) {
        t += (retry > 0 ? retry : 1);
        s.drop_waiters({ck - 1, ck - 2});
        continue;
      }
      t += g.quota_ms;
      s.release(g.pod, g.quota_ms, t);
      used[g.pod == "B"] += g.quota_ms;
      s.drop_waiters({ck - 1, ck - 2});
    }
    double tot = used[0] + used[1];
    printf("  %.2f %.2f %.2f %.2f -> %.3f %.3f  (util %.2f)\n", m[0],
           m[1], m[2], m[3], used[0] / tot, used[1] / tot, tot / 120000);
  }
  return 0;
}
