// kubeshare-amd native layer — shared wire protocol & helpers.
//
// The isolation data path is:
//   libhiphook.so (in-container, LD_PRELOAD)
//     --TCP/UDS--> pod-mgr (per-pod, trusted identity stamp)
//     --TCP/UDS--> gpu-schd (per-GPU sliding-window token scheduler)
//
// This mirrors the reference's Gemini topology (SURVEY.md §2.2:
// libgemhook.so.1 <-> gem-pmgr <-> gem-schd; env contract
// docker/kubeshare-gemini-scheduler/launcher.py:13-20) but the protocol
// is re-derived from first principles: newline-delimited text so every
// daemon is unit-testable on a CPU-only box with a Python socket.
//
// Messages (client -> scheduler):
//   REQ <pod> <hint_ms>\n     ask for a time-slice token. The reply
//                             (GRANT <quota_ms>\n) may be delayed
//                             arbitrarily — the delay IS the throttle.
//   RET <pod> <used_ms>\n     return the token, reporting measured use.
//                             Reply: OK\n
//   MEM <pod> <bytes>\n       report current device-memory footprint
//                             (observability only). Reply: OK\n
//   STATS\n                   reply: one line of JSON.
//
// Replies are self-describing (first token), so the link is full-duplex:
// a pod-mgr can forward one client's RET while another client's REQ is
// still waiting for its GRANT. Per connection, GRANTs answer REQs in
// FIFO order and OKs answer RET/MEM in FIFO order.
#pragma once

#include <arpa/inet.h>
#include <errno.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <stdarg.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <time.h>
#include <unistd.h>

#include <string>
#include <vector>

namespace ks {

inline double now_ms() {
  struct timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return ts.tv_sec * 1e3 + ts.tv_nsec * 1e-6;
}

// ---------------------------------------------------------------- sockets

// Connect to host:port (TCP, TCP_NODELAY) or, when `host` starts with
// '/', to that Unix-domain socket path. Returns fd or -1.
inline int connect_to(const char* host, int port, double timeout_ms = 10000.0) {
  double deadline = now_ms() + timeout_ms;
  for (;;) {
    int fd = -1;
    if (host[0] == '/') {
      fd = ::socket(AF_UNIX, SOCK_STREAM, 0);
      if (fd < 0) return -1;
      struct sockaddr_un addr;
      memset(&addr, 0, sizeof(addr));
      addr.sun_family = AF_UNIX;
      snprintf(addr.sun_path, sizeof(addr.sun_path), "%s", host);
      if (::connect(fd, (struct sockaddr*)&addr, sizeof(addr)) == 0) return fd;
    } else {
      fd = ::socket(AF_INET, SOCK_STREAM, 0);
      if (fd < 0) return -1;
      int one = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      struct sockaddr_in addr;
      memset(&addr, 0, sizeof(addr));
      addr.sin_family = AF_INET;
      addr.sin_port = htons((uint16_t)port);
      if (inet_pton(AF_INET, host, &addr.sin_addr) != 1) {
        ::close(fd);
        return -1;
      }
      if (::connect(fd, (struct sockaddr*)&addr, sizeof(addr)) == 0) return fd;
    }
    ::close(fd);
    if (now_ms() >= deadline) return -1;
    usleep(50 * 1000);  // daemon may still be starting; retry
  }
}

// Listening socket: TCP on `port` when path==nullptr, else UDS at path.
inline int listen_on(const char* path, int port, int backlog = 128) {
  int fd;
  if (path && path[0]) {
    ::unlink(path);
    fd = ::socket(AF_UNIX, SOCK_STREAM, 0);
    if (fd < 0) return -1;
    struct sockaddr_un addr;
    memset(&addr, 0, sizeof(addr));
    addr.sun_family = AF_UNIX;
    snprintf(addr.sun_path, sizeof(addr.sun_path), "%s", path);
    if (::bind(fd, (struct sockaddr*)&addr, sizeof(addr)) < 0) {
      ::close(fd);
      return -1;
    }
  } else {
    fd = ::socket(AF_INET, SOCK_STREAM, 0);
    if (fd < 0) return -1;
    int one = 1;
    setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    struct sockaddr_in addr;
    memset(&addr, 0, sizeof(addr));
    addr.sin_family = AF_INET;
    addr.sin_port = htons((uint16_t)port);
    addr.sin_addr.s_addr = htonl(INADDR_ANY);
    if (::bind(fd, (struct sockaddr*)&addr, sizeof(addr)) < 0) {
      ::close(fd);
      return -1;
    }
  }
  if (::listen(fd, backlog) < 0) {
    ::close(fd);
    return -1;
  }
  // poll-loop daemons drain accept() until EAGAIN — must not block
  fcntl(fd, F_SETFL, fcntl(fd, F_GETFL, 0) | O_NONBLOCK);
  return fd;
}

inline bool send_all(int fd, const char* buf, size_t len) {
  while (len > 0) {
    ssize_t n = ::send(fd, buf, len, MSG_NOSIGNAL);
    if (n < 0) {
      if (errno == EINTR) continue;
      return false;
    }
    buf += n;
    len -= (size_t)n;
  }
  return true;
}

inline bool send_line(int fd, const std::string& line) {
  std::string s = line;
  if (s.empty() || s.back() != '\n') s.push_back('\n');
  return send_all(fd, s.data(), s.size());
}

// Blocking read of one '\n'-terminated line (small-message use: hook
// client). Returns false on EOF/error.
inline bool recv_line(int fd, std::string& out) {
  out.clear();
  char c;
  for (;;) {
    ssize_t n = ::recv(fd, &c, 1, 0);
    if (n == 0) return false;
    if (n < 0) {
      if (errno == EINTR) continue;
      return false;
    }
    if (c == '\n') return true;
    out.push_back(c);
    if (out.size() > 1 << 20) return false;
  }
}

// Per-connection receive buffer for poll-loop daemons.
struct LineBuffer {
  std::string buf;
  // Append freshly received bytes, then pop complete lines.
  void feed(const char* data, size_t len) { buf.append(data, len); }
  bool pop(std::string& line) {
    size_t p = buf.find('\n');
    if (p == std::string::npos) return false;
    line.assign(buf, 0, p);
    buf.erase(0, p + 1);
    return true;
  }
};

inline std::vector<std::string> split_ws(const std::string& s) {
  std::vector<std::string> out;
  size_t i = 0;
  while (i < s.size()) {
    while (i < s.size() && (s[i] == ' ' || s[i] == '\t')) i++;
    size_t j = i;
    while (j < s.size() && s[j] != ' ' && s[j] != '\t') j++;
    if (j > i) out.emplace_back(s.substr(i, j - i));
    i = j;
  }
  return out;
}

// ------------------------------------------------- per-UUID config file
// Format (reference pkg/config/query.go:70-84):
//   n\n
//   <namespace>/<name> <limit> <request> <memory>\n  x n
struct PodQuota {
  std::string pod;
  double limit = 1.0;
  double request = 0.0;
  long long memory = 0;
  // optional trailing fields: gang group (bare word — members are
  // co-granted so a collective in one rank never spins on a
  // token-starved peer) and q=<ms> (latency class: per-pod
  // lease-length override, clamped to [min_quota, base_quota])
  std::string group;
  double lease_ms = 0.0;
};

inline bool parse_gpu_config(FILE* f, std::vector<PodQuota>& out) {
  out.clear();
  char line[4096];
  if (!fgets(line, sizeof(line), f)) return false;
  int n = atoi(line);
  for (int i = 0; i < n; i++) {
    if (!fgets(line, sizeof(line), f)) return false;
    PodQuota q;
    char pod[2048];
    char e5[1024] = "";
    char e6[1024] = "";
    int got = sscanf(line, "%2047s %lf %lf %lld %1023s %1023s", pod,
                     &q.limit, &q.request, &q.memory, e5, e6);
    if (got < 4) return false;
    q.pod = pod;
    for (const char* extra : {e5, e6}) {
      if (!extra[0]) continue;
      if (extra[0] == 'q' && extra[1] == '=')
        q.lease_ms = atof(extra + 2);
      else
        q.group = extra;
    }
    out.push_back(q);
  }
  return true;
}

inline bool load_gpu_config(const char* path, std::vector<PodQuota>& out) {
  FILE* f = fopen(path, "r");
  if (!f) return false;
  bool ok = parse_gpu_config(f, out);
  fclose(f);
  return ok;
}

// ------------------------------------------------------------------ log
inline void logf(FILE* sink, const char* comp, const char* fmt, ...) {
  if (!sink) return;
  struct timespec ts;
  clock_gettime(CLOCK_REALTIME, &ts);
  struct tm tmv;
  localtime_r(&ts.tv_sec, &tmv);
  char head[64];
  strftime(head, sizeof(head), "%F %T", &tmv);
  fprintf(sink, "%s.%03ld %s: ", head, ts.tv_nsec / 1000000, comp);
  va_list ap;
  va_start(ap, fmt);
  vfprintf(sink, fmt, ap);
  va_end(ap);
  fputc('\n', sink);
  fflush(sink);
}

}  // namespace ks
