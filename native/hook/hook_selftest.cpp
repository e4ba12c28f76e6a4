// hook_selftest — CPU-only driver for the token-gate state machine.
//
// Simulates one GPU-bound pod: acquire a lease, "occupy the GPU" for the
// lease (sleep — on a real GPU the exclusivity is physical; here wall
// time models it), return it, repeat. Lets tests/test_native_daemons.py
// verify the full hook->pod-mgr->gpu-schd chain and the request:limit
// split on a box with no GPU at all — the fixture layer the reference
// never had.
//
//   hook_selftest <host|uds> <port> <pod> <duration_ms> [stats]
//   hook_selftest <host|uds> <port> <pod> <duration_ms> gate <active_ms> <idle_ms>
//
// The second form drives the REAL TokenGate (the object libhiphook uses)
// with a stub drain, alternating bursts of simulated dispatches with
// idle phases — exercising lease renewal and the idle-release watchdog.
//
// Prints one line: "DONE <pod> <leases> <granted_ms>" and, with `stats`,
// the scheduler's STATS JSON line.
#include "token_client.hpp"

using namespace ks;

static int run_gate_mode(const char* host, int port, const std::string& pod,
                         double duration, double active_ms, double idle_ms) {
  static TokenGate gate;
  gate.init(host, port, pod, [](void*) { /* GPU already drained */ },
            nullptr);
  double t0 = now_ms();
  while (now_ms() - t0 < duration) {
    double burst_end = now_ms() + active_ms;
    while (now_ms() < burst_end && now_ms() - t0 < duration) {
      gate.ensure();  // a "kernel launch" every ~1 ms
      usleep(1000);
    }
    if (idle_ms > 0) usleep((useconds_t)(idle_ms * 1000));
  }
  gate.relinquish();
  printf("DONE %s %lld %.1f idle_releases=%lld\n", pod.c_str(),
         gate.leases(), gate.used_ms_total(), gate.idle_releases_);
  return 0;
}

int main(int argc, char** argv) {
  if (argc < 5) {
    fprintf(stderr, "usage: hook_selftest <host> <port> <pod> <duration_ms> "
                    "[stats | gate <active_ms> <idle_ms>]\n");
    return 2;
  }
  const char* host = argv[1];
  int port = atoi(argv[2]);
  std::string pod = argv[3];
  double duration = atof(argv[4]);
  bool want_stats = argc >= 6 && strcmp(argv[5], "stats") == 0;
  if (argc >= 8 && strcmp(argv[5], "gate") == 0)
    return run_gate_mode(host, port, pod, duration, atof(argv[6]),
                         atof(argv[7]));

  TokenClient client;
  client.configure(host, port, pod);

  double t0 = now_ms();
  double granted = 0.0;
  long long leases = 0;
  while (now_ms() - t0 < duration) {
    double quota = client.acquire(0.0);
    if (quota <= 0.0) {
      fprintf(stderr, "hook_selftest: acquire failed\n");
      return 1;
    }
    leases++;
    // occupy for the lease (a GPU-bound pod's wall == busy time)
    double start = now_ms();
    double hold = std::min(quota, t0 + duration - start + 1.0);
    if (hold > 0) usleep((useconds_t)(hold * 1000));
    double used = now_ms() - start;
    granted += used;
    client.return_token(used);
  }
  printf("DONE %s %lld %.1f\n", pod.c_str(), leases, granted);

  if (want_stats) {
    int fd = connect_to(host, port, 5000.0);
    if (fd >= 0) {
      send_line(fd, "STATS");
      std::string reply;
      while (recv_line(fd, reply)) {
        if (!reply.empty() && reply[0] == '{') {
          printf("%s\n", reply.c_str());
          break;
        }
      }
      close(fd);
    }
  }
  return 0;
}
