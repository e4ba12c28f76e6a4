// gpu-schd — per-GPU token scheduler daemon (MI355X-native gem-schd).
//
// CLI contract matches the reference launcher's invocation of gem-schd
// (docker/kubeshare-gemini-scheduler/launcher.py:27-31, defaults 77-80):
//   gpu-schd -p <config dir> -f <config file (=GPU UUID)> -P <port>
//            [-q base_quota_ms=300] [-m min_quota_ms=20]
//            [-w window_ms=10000] [-U <unix socket path>] [-l <log file>]
//
// Single-threaded poll(2) event loop; the scheduling policy lives in
// token_sched.hpp. The per-UUID config file is hot-reloaded via
// inotify(7) (the reference launcher watched the same directory,
// launcher.py:89-98).
#include <getopt.h>
#include <poll.h>
#include <signal.h>
#include <sys/inotify.h>

#include <map>
#include <memory>
#include <set>

#include "busy_sampler.hpp"
#include "token_sched.hpp"

using namespace ks;

namespace {

struct Client {
  int fd;
  long long conn_id;            // unique (fds are reused by the kernel)
  LineBuffer rx;
  std::set<long long> pending;  // outstanding REQ cookies of this conn
};

FILE* g_log = stderr;

}  // namespace

int main(int argc, char** argv) {
  std::string config_dir, config_file, uds_path, log_path;
  int port = 0;
  int gpu_index = -1;  // -d: enable the server-side busy sampler
  double base_q = 300.0, min_q = 20.0, window = 10000.0;

  int opt;
  while ((opt = getopt(argc, argv, "p:f:P:q:m:w:U:l:d:")) != -1) {
    switch (opt) {
      case 'p': config_dir = optarg; break;
      case 'f': config_file = optarg; break;
      case 'P': port = atoi(optarg); break;
      case 'q': base_q = atof(optarg); break;
      case 'm': min_q = atof(optarg); break;
      case 'w': window = atof(optarg); break;
      case 'U': uds_path = optarg; break;
      case 'l': log_path = optarg; break;
      case 'd': gpu_index = atoi(optarg); break;
      default:
        fprintf(stderr,
                "usage: gpu-schd -p <dir> -f <file> -P <port> [-q ms] [-m ms] "
                "[-w ms] [-U uds] [-l log] [-d gpu_index]\n");
        return 2;
    }
  }
  if (port == 0 && uds_path.empty()) {
    fprintf(stderr, "gpu-schd: need -P <port> or -U <uds>\n");
    return 2;
  }
  if (!log_path.empty()) {
    FILE* f = fopen(log_path.c_str(), "a");
    if (f) g_log = f;
  }
  signal(SIGPIPE, SIG_IGN);

  TokenScheduler sched(base_q, min_q, window);

  BusySampler sampler;
  if (gpu_index >= 0) {
    if (sampler.init(gpu_index)) {
      logf(g_log, "gpu-schd", "busy sampler on (gpu %d): leases charged "
           "sampled GPU-busy, not wall", gpu_index);
    } else {
      logf(g_log, "gpu-schd", "busy sampler unavailable (gpu %d): "
           "falling back to wall/RET accounting", gpu_index);
    }
  }

  // Order matters: install the config WATCH before anyone can observe
  // us as ready (the listen socket), then load — a config written
  // between an initial load and a later watch would be missed forever.
  std::string config_path;
  int ino_fd = -1;
  if (!config_dir.empty() && !config_file.empty()) {
    if (config_dir.back() != '/') config_dir.push_back('/');
    config_path = config_dir + config_file;
    ino_fd = inotify_init1(IN_NONBLOCK);
    if (ino_fd >= 0) {
      (void)inotify_add_watch(ino_fd, config_dir.c_str(),
                              IN_CLOSE_WRITE | IN_MOVED_TO | IN_CREATE);
    }
    std::vector<PodQuota> quotas;
    if (load_gpu_config(config_path.c_str(), quotas)) {
      sched.set_config(quotas, now_ms());
      logf(g_log, "gpu-schd", "loaded %zu pod quotas from %s", quotas.size(),
           config_path.c_str());
    }
  }

  int listen_fd = listen_on(uds_path.empty() ? nullptr : uds_path.c_str(), port);
  if (listen_fd < 0) {
    logf(g_log, "gpu-schd", "FATAL: cannot listen (%s port %d): %s",
         uds_path.c_str(), port, strerror(errno));
    return 1;
  }
  logf(g_log, "gpu-schd", "listening (%s port %d) q=%.0f m=%.0f w=%.0f",
       uds_path.c_str(), port, base_q, min_q, window);

  std::map<int, std::unique_ptr<Client>> clients;
  std::map<long long, int> cookie2fd;  // outstanding REQ cookie -> client fd
  long long next_cookie = 1;
  long long next_conn = 1;
  double wake_in = -1.0;  // ms until a capped waiter may become eligible

  auto flush_grants = [&]() {
    // loop until no further grant is possible: gang co-granting means
    // several same-group members can be granted back-to-back
    for (;;) {
      Grant g;
      double retry = 0.0;
      if (!sched.schedule(now_ms(), &g, &retry)) {
        wake_in = retry > 0.0 ? retry : -1.0;
        break;
      }
      auto it = cookie2fd.find(g.cookie);
      if (it == cookie2fd.end()) {
        sched.force_release(g.pod, now_ms());
        continue;
      }
      auto cl = clients.find(it->second);
      if (cl != clients.end()) {
        cl->second->pending.erase(g.cookie);
        // remember the owning CONNECTION so only its death frees the
        // grant (a same-pod sibling connection dying must not)
        sched.set_holder_owner(g.pod, cl->second->conn_id);
      }
      char line[64];
      snprintf(line, sizeof(line), "GRANT %.3f", g.quota_ms);
      if (!send_line(it->second, line)) {
        // client went away between REQ and GRANT: free the token
        sched.force_release(g.pod, now_ms());
      }
      cookie2fd.erase(it);
    }
  };

  for (;;) {
    std::vector<struct pollfd> pfds;
    pfds.push_back({listen_fd, POLLIN, 0});
    if (ino_fd >= 0) pfds.push_back({ino_fd, POLLIN, 0});
    for (auto& kv : clients) pfds.push_back({kv.first, POLLIN, 0});

    int timeout = 200;  // liveness-check cadence
    if (wake_in > 0.0) timeout = (int)std::min(wake_in, 200.0) + 1;
    // the sampler needs a short tick to attribute busy time accurately
    if (sampler.active() && sched.n_holders() > 0 && timeout > 5)
      timeout = 5;
    ::poll(pfds.data(), pfds.size(), timeout);
    wake_in = -1.0;
    if (sampler.active()) sched.add_busy(sampler.poll(), now_ms());

    size_t idx = 0;
    // new connections
    if (pfds[idx++].revents & POLLIN) {
      int cfd;
      while ((cfd = ::accept(listen_fd, nullptr, nullptr)) >= 0) {
        int one = 1;
        setsockopt(cfd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
        auto c = std::make_unique<Client>();
        c->fd = cfd;
        c->conn_id = next_conn++;
        clients[cfd] = std::move(c);
        if (clients.size() >= 1024) break;
      }
    }
    // config reload
    if (ino_fd >= 0 && (pfds[idx++].revents & POLLIN)) {
      char evbuf[4096];
      ssize_t n = read(ino_fd, evbuf, sizeof(evbuf));
      bool ours = false;
      for (ssize_t off = 0; off < n;) {
        auto* ev = (struct inotify_event*)(evbuf + off);
        if (ev->len && config_file == ev->name) ours = true;
        off += sizeof(struct inotify_event) + ev->len;
      }
      if (ours && !config_path.empty()) {
        std::vector<PodQuota> quotas;
        if (load_gpu_config(config_path.c_str(), quotas)) {
          sched.set_config(quotas, now_ms());
          logf(g_log, "gpu-schd", "config reloaded: %zu pods", quotas.size());
        }
      }
    }

    // client traffic
    std::vector<int> dead;
    for (auto& kv : clients) {
      Client& c = *kv.second;
      struct pollfd* p = nullptr;
      for (size_t k = idx; k < pfds.size(); k++)
        if (pfds[k].fd == c.fd) {
          p = &pfds[k];
          break;
        }
      if (!p || !(p->revents & (POLLIN | POLLHUP | POLLERR))) continue;
      char buf[4096];
      ssize_t n = ::recv(c.fd, buf, sizeof(buf), 0);
      if (n <= 0) {
        dead.push_back(c.fd);
        continue;
      }
      c.rx.feed(buf, (size_t)n);
      std::string line;
      while (c.rx.pop(line)) {
        auto tok = split_ws(line);
        if (tok.empty()) continue;
        double now = now_ms();
        if (tok[0] == "REQ" && tok.size() >= 2) {
          double hint = tok.size() >= 3 ? atof(tok[2].c_str()) : 0.0;
          long long ck = next_cookie++;
          c.pending.insert(ck);
          cookie2fd[ck] = c.fd;
          sched.request(tok[1], ck, hint, now);
        } else if (tok[0] == "RET" && tok.size() >= 3) {
          sched.release(tok[1], atof(tok[2].c_str()), now);
          send_line(c.fd, "OK");
        } else if (tok[0] == "MEM" && tok.size() >= 3) {
          sched.account(tok[1]).mem_bytes = atoll(tok[2].c_str());
          send_line(c.fd, "OK");
        } else if (tok[0] == "STATS") {
          send_line(c.fd, sched.stats_json(now));
        } else {
          send_line(c.fd, "ERR unknown");
        }
      }
    }
    for (int fd : dead) {
      Client& c = *clients[fd];
      // cancel outstanding REQs from this conn; free exactly the
      // grants THIS connection owns (not everything under its pod name
      // — a live sibling connection may hold a granted token)
      sched.drop_waiters(c.pending);
      for (auto it = cookie2fd.begin(); it != cookie2fd.end();)
        it = (it->second == fd) ? cookie2fd.erase(it) : std::next(it);
      sched.force_release_owned(c.conn_id, now_ms());
      ::close(fd);
      clients.erase(fd);
    }

    sched.check_revoke(now_ms());
    flush_grants();
  }
  return 0;
}
