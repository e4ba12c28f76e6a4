// pod-mgr — per-pod token proxy (MI355X-native gem-pmgr).
//
// One instance per sharing pod, spawned by the node launcher as the
// podmanagerport/<uuid> file changes (reference launcher.py:34-67). Env
// contract is the reference's (launcher.py:13-20):
//   SCHEDULER_IP / SCHEDULER_PORT   upstream gpu-schd
//   POD_MANAGER_IP / POD_MANAGER_PORT  where we listen for hook clients
//   POD_NAME                        "<namespace>/<name>"
// Optional: SCHEDULER_UDS / POD_MANAGER_UDS (unix sockets, preferred on
// one node — drops the reference's hostNetwork requirement).
//
// Security property (why this hop exists at all): the pod identity sent
// upstream is stamped from OUR env, not from what the in-container hook
// claims — a malicious container cannot appropriate another pod's
// quota. The link is full-duplex: replies are matched by type FIFO
// (GRANT->REQ, OK->RET/MEM, JSON->STATS) so one in-pod process can RET
// while another's REQ is still pending (see protocol.hpp).
#include <poll.h>
#include <signal.h>

#include <deque>
#include <map>
#include <memory>

#include "../common/protocol.hpp"

using namespace ks;

namespace {

struct Client {
  int fd;
  LineBuffer rx;
};

}  // namespace

int main() {
  signal(SIGPIPE, SIG_IGN);
  const char* sched_ip = getenv("SCHEDULER_IP");
  const char* sched_port = getenv("SCHEDULER_PORT");
  const char* sched_uds = getenv("SCHEDULER_UDS");
  const char* mgr_port = getenv("POD_MANAGER_PORT");
  const char* mgr_uds = getenv("POD_MANAGER_UDS");
  const char* pod_name_env = getenv("POD_NAME");
  std::string pod = pod_name_env ? pod_name_env : "unknown/unknown";

  FILE* log = stderr;
  if (const char* lp = getenv("POD_MANAGER_LOG")) {
    FILE* f = fopen(lp, "a");
    if (f) log = f;
  }

  int up = -1;
  auto connect_up = [&]() -> int {
    if (sched_uds && sched_uds[0])
      return connect_to(sched_uds, 0, 30000.0);
    return connect_to(sched_ip && sched_ip[0] ? sched_ip : "127.0.0.1",
                      sched_port ? atoi(sched_port) : 0, 30000.0);
  };
  up = connect_up();
  if (up < 0) {
    logf(log, "pod-mgr", "FATAL: cannot reach gpu-schd");
    return 1;
  }

  // Listen on BOTH when both are configured: UDS through the
  // /kubeshare/sock hostPath is the default transport (unreachable
  // from pods that don't mount it — no cross-pod quota-burn, no
  // hostNetwork requirement); TCP stays as the fallback for setups
  // without the shared mount.
  int uds_fd = -1, tcp_fd = -1;
  if (mgr_uds && mgr_uds[0]) {
    uds_fd = listen_on(mgr_uds, 0);
    if (uds_fd < 0)
      logf(log, "pod-mgr", "WARN: cannot listen on uds %s: %s", mgr_uds,
           strerror(errno));
  }
  if (mgr_port && atoi(mgr_port) > 0) {
    tcp_fd = listen_on(nullptr, atoi(mgr_port));
    if (tcp_fd < 0)
      logf(log, "pod-mgr", "WARN: cannot listen on port %s: %s", mgr_port,
           strerror(errno));
  }
  if (uds_fd < 0 && tcp_fd < 0) {
    logf(log, "pod-mgr", "FATAL: cannot listen on %s/%s",
         mgr_uds ? mgr_uds : "-", mgr_port ? mgr_port : "-");
    return 1;
  }
  logf(log, "pod-mgr", "pod=%s listening (uds=%s port=%s)", pod.c_str(),
       uds_fd >= 0 ? mgr_uds : "-", tcp_fd >= 0 ? mgr_port : "-");

  std::map<int, std::unique_ptr<Client>> clients;
  LineBuffer up_rx;
  std::deque<int> grant_q;  // client fds awaiting GRANT (FIFO)
  std::deque<int> ok_q;     // client fds awaiting OK
  std::deque<int> stats_q;  // client fds awaiting stats JSON

  auto drop_from = [](std::deque<int>& q, int fd) {
    for (auto& x : q)
      if (x == fd) x = -1;  // keep FIFO positions; -1 = discard reply
  };

  const int listeners[2] = {uds_fd, tcp_fd};
  for (;;) {
    std::vector<struct pollfd> pfds;
    for (int lf : listeners)
      if (lf >= 0) pfds.push_back({lf, POLLIN, 0});
    size_t nlisten = pfds.size();
    pfds.push_back({up, POLLIN, 0});
    for (auto& kv : clients) pfds.push_back({kv.first, POLLIN, 0});
    ::poll(pfds.data(), pfds.size(), 1000);

    for (size_t li = 0; li < nlisten; li++) {
      if (!(pfds[li].revents & POLLIN)) continue;
      int cfd;
      while ((cfd = ::accept(pfds[li].fd, nullptr, nullptr)) >= 0) {
        int one = 1;
        setsockopt(cfd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
        auto c = std::make_unique<Client>();
        c->fd = cfd;
        clients[cfd] = std::move(c);
      }
    }

    // upstream replies -> route by type FIFO
    if (pfds[nlisten].revents & (POLLIN | POLLHUP | POLLERR)) {
      char buf[4096];
      ssize_t n = ::recv(up, buf, sizeof(buf), 0);
      if (n <= 0) {
        logf(log, "pod-mgr", "upstream lost; exiting");
        return 1;  // node launcher restarts us; hook clients reconnect
      }
      up_rx.feed(buf, (size_t)n);
      std::string line;
      while (up_rx.pop(line)) {
        std::deque<int>* q = nullptr;
        if (line.rfind("GRANT", 0) == 0) q = &grant_q;
        else if (line.rfind("OK", 0) == 0) q = &ok_q;
        else if (!line.empty() && line[0] == '{') q = &stats_q;
        if (!q || q->empty()) continue;
        int cfd = q->front();
        q->pop_front();
        if (cfd >= 0 && clients.count(cfd)) {
          if (line.rfind("GRANT", 0) == 0 && !send_line(cfd, line)) {
            // client died as its grant arrived: give the token back
            send_line(up, "RET " + pod + " 0");
            ok_q.push_back(-1);
          } else if (line.rfind("GRANT", 0) != 0) {
            send_line(cfd, line);
          }
        } else if (cfd < 0 && line.rfind("GRANT", 0) == 0) {
          send_line(up, "RET " + pod + " 0");
          ok_q.push_back(-1);
        }
      }
    }

    // client requests -> stamp identity, forward upstream
    std::vector<int> dead;
    for (size_t k = nlisten + 1; k < pfds.size(); k++) {
      if (!(pfds[k].revents & (POLLIN | POLLHUP | POLLERR))) continue;
      int cfd = pfds[k].fd;
      auto it = clients.find(cfd);
      if (it == clients.end()) continue;
      Client& c = *it->second;
      char buf[4096];
      ssize_t n = ::recv(cfd, buf, sizeof(buf), 0);
      if (n <= 0) {
        dead.push_back(cfd);
        continue;
      }
      c.rx.feed(buf, (size_t)n);
      std::string line;
      while (c.rx.pop(line)) {
        auto tok = split_ws(line);
        if (tok.empty()) continue;
        if (tok[0] == "REQ") {
          std::string hint = tok.size() >= 3 ? tok[2] : "0";
          send_line(up, "REQ " + pod + " " + hint);
          grant_q.push_back(cfd);
        } else if (tok[0] == "RET") {
          std::string used = tok.size() >= 3 ? tok[2] : "0";
          send_line(up, "RET " + pod + " " + used);
          ok_q.push_back(cfd);
        } else if (tok[0] == "MEM") {
          std::string bytes = tok.size() >= 3 ? tok[2] : "0";
          send_line(up, "MEM " + pod + " " + bytes);
          ok_q.push_back(cfd);
        } else if (tok[0] == "STATS") {
          send_line(up, "STATS");
          stats_q.push_back(cfd);
        }
      }
    }
    for (int fd : dead) {
      drop_from(grant_q, fd);
      drop_from(ok_q, fd);
      drop_from(stats_q, fd);
      ::close(fd);
      clients.erase(fd);
      // if this client held the token it will be revoked upstream by
      // gpu-schd's liveness check; nothing to do here
    }
  }
}
