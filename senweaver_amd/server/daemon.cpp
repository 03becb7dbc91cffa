// senweaver_daemon — native JSON-RPC streaming control plane.
//
// The reference splits LLM transport across processes: the renderer posts
// requests over the IPC channel 'senweaver-channel-llmMessage' and the
// Electron main process streams request-id-tagged events back
// (electron-main/sendLLMMessageChannel.ts: listen onText_/onFinalMessage_/
// onError_, call sendLLMMessage/abort/list; per-request abortRef map).
//
// This daemon is the MI355X-native equivalent: clients connect over a Unix
// domain socket and speak newline-delimited JSON:
//   -> {"method":"sendLLMMessage","requestId":"r1", ...payload}
//   -> {"method":"abort","requestId":"r1"}
//   -> {"method":"list"}
//   <- {"event":"onText","requestId":"r1","fullText":...}    (streamed)
//   <- {"event":"onFinalMessage","requestId":"r1",...}
//   <- {"event":"onError","requestId":"r1","message":...}
//
// The daemon owns connection routing, the request-id -> client map, the
// instant client-side abort fast path (an aborted id suppresses further
// events immediately, before the engine has even seen the abort — the same
// no-round-trip semantics as sendLLMMessageService.ts:142-146), and engine
// lifecycle; the Python engine worker (engine_worker.py, spawned as a child
// on the same box, GPU-side) does the model work.  Payloads transit the
// daemon opaquely — only "method"/"requestId"/"event" are inspected.

#include <arpa/inet.h>
#include <errno.h>
#include <fcntl.h>
#include <poll.h>
#include <signal.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <sys/wait.h>
#include <time.h>
#include <unistd.h>

#include <map>
#include <set>
#include <string>
#include <vector>

namespace {

// Minimal extraction of a top-level string field from a JSON object line.
// Handles escaped quotes; payloads are otherwise passed through opaquely.
std::string json_field(const std::string& line, const std::string& key) {
  const std::string pat = "\"" + key + "\"";
  size_t p = line.find(pat);
  if (p == std::string::npos) return "";
  p = line.find(':', p + pat.size());
  if (p == std::string::npos) return "";
  ++p;
  while (p < line.size() && (line[p] == ' ' || line[p] == '\t')) ++p;
  if (p >= line.size() || line[p] != '"') return "";
  ++p;
  std::string out;
  while (p < line.size()) {
    char c = line[p];
    if (c == '\\' && p + 1 < line.size()) {
      out.push_back(line[p + 1]);
      p += 2;
      continue;
    }
    if (c == '"') break;
    out.push_back(c);
    ++p;
  }
  return out;
}

struct Client {
  int fd;
  std::string inbuf;
  std::string outbuf;  // pending events not yet accepted by the socket
};

// A consumer that stops reading gets its events buffered up to this cap and
// is then dropped (its requests aborted) — the daemon never buffers
// unboundedly for a stalled client.
constexpr size_t kMaxClientOutbuf = 8u << 20;  // 8 MiB

class Daemon {
 public:
  Daemon(std::string socket_path, std::vector<std::string> worker_argv)
      : socket_path_(std::move(socket_path)), worker_argv_(std::move(worker_argv)) {}

  int run() {
    signal(SIGPIPE, SIG_IGN);
    if (!spawn_worker()) return 1;
    listen_fd_ = make_listen_socket();
    if (listen_fd_ < 0) return 1;
    fprintf(stderr, "[daemon] listening on %s\n", socket_path_.c_str());
    loop();
    return 0;
  }

 private:
  std::string socket_path_;
  std::vector<std::string> worker_argv_;
  int listen_fd_ = -1;
  int worker_in_ = -1;   // write requests to worker stdin
  int worker_out_ = -1;  // read events from worker stdout
  pid_t worker_pid_ = -1;
  std::string worker_buf_;
  std::map<int, Client> clients_;
  std::map<std::string, int> request_client_;  // requestId -> client fd
  std::set<std::string> aborted_;
  time_t last_spawn_ = 0;
  int fast_crashes_ = 0;  // crash-loop breaker (worker dead < 20 s after spawn)

  bool spawn_worker() {
    last_spawn_ = time(nullptr);
    int in_pipe[2], out_pipe[2];
    if (pipe(in_pipe) || pipe(out_pipe)) return false;
    worker_pid_ = fork();
    if (worker_pid_ < 0) return false;
    if (worker_pid_ == 0) {
      dup2(in_pipe[0], 0);
      dup2(out_pipe[1], 1);
      close(in_pipe[1]);
      close(out_pipe[0]);
      std::vector<char*> argv;
      for (auto& a : worker_argv_) argv.push_back(const_cast<char*>(a.c_str()));
      argv.push_back(nullptr);
      execvp(argv[0], argv.data());
      perror("execvp engine worker");
      _exit(127);
    }
    close(in_pipe[0]);
    close(out_pipe[1]);
    worker_in_ = in_pipe[1];
    worker_out_ = out_pipe[0];
    return true;
  }

  int make_listen_socket() {
    int fd = socket(AF_UNIX, SOCK_STREAM, 0);
    if (fd < 0) return -1;
    sockaddr_un addr{};
    addr.sun_family = AF_UNIX;
    unlink(socket_path_.c_str());
    snprintf(addr.sun_path, sizeof(addr.sun_path), "%s", socket_path_.c_str());
    if (bind(fd, (sockaddr*)&addr, sizeof(addr)) || listen(fd, 16)) {
      perror("bind/listen");
      close(fd);
      return -1;
    }
    return fd;
  }

  static bool write_all(int fd, const std::string& s) {
    size_t off = 0;
    while (off < s.size()) {
      ssize_t n = write(fd, s.data() + off, s.size() - off);
      if (n <= 0) {
        if (n < 0 && errno == EINTR) continue;
        return false;
      }
      off += (size_t)n;
    }
    return true;
  }

  // Queue an event line for a client; non-blocking flush.  Returns false if
  // the client is gone or stalled past the buffer cap (caller drops it).
  bool send_to_client(int fd, const std::string& line) {
    auto it = clients_.find(fd);
    if (it == clients_.end()) return false;
    Client& c = it->second;
    c.outbuf += line;
    c.outbuf += '\n';
    if (!flush_client(c)) return false;
    return c.outbuf.size() <= kMaxClientOutbuf;
  }

  static bool flush_client(Client& c) {
    while (!c.outbuf.empty()) {
      ssize_t n = write(c.fd, c.outbuf.data(), c.outbuf.size());
      if (n > 0) {
        c.outbuf.erase(0, (size_t)n);
        continue;
      }
      if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) return true;
      if (n < 0 && errno == EINTR) continue;
      return false;  // broken pipe / closed
    }
    return true;
  }

  void handle_client_line(int fd, const std::string& line) {
    const std::string method = json_field(line, "method");
    const std::string rid = json_field(line, "requestId");
    if (method == "sendLLMMessage") {
      if (rid.empty()) {
        send_to_client(fd, "{\"event\":\"onError\",\"message\":\"missing requestId\"}");
        return;
      }
      request_client_[rid] = fd;
      aborted_.erase(rid);
      write_all(worker_in_, line + "\n");
    } else if (method == "abort") {
      // instant client-side abort: suppress events immediately, then tell
      // the engine so it stops decoding
      aborted_.insert(rid);
      request_client_.erase(rid);
      write_all(worker_in_, line + "\n");
      send_to_client(fd, "{\"event\":\"onAbort\",\"requestId\":\"" + rid + "\"}");
    } else if (method == "list" || method == "ping" || method == "stats") {
      write_all(worker_in_, line + "\n");
      request_client_[method] = fd;  // single in-flight list/ping per method
    } else if (method == "shutdown") {
      send_to_client(fd, "{\"event\":\"shuttingDown\"}");
      shutdown_all();
      exit(0);
    } else {
      send_to_client(fd, "{\"event\":\"onError\",\"message\":\"unknown method\"}");
    }
  }

  void handle_worker_line(const std::string& line) {
    const std::string rid = json_field(line, "requestId");
    const std::string ev = json_field(line, "event");
    const bool terminal =
        ev == "onFinalMessage" || ev == "onError" || ev == "onAbort";
    // aborted ids: suppress events but reap the tombstone once the worker
    // acknowledges the end of the request, so aborted_ stays bounded
    if (!rid.empty() && aborted_.count(rid)) {
      if (terminal) aborted_.erase(rid);
      return;
    }
    std::string key = rid;
    if (key.empty())
      key = ev == "listResult" ? "list"
            : ev == "pong" ? "ping"
            : ev == "statsResult" ? "stats" : "";
    auto it = request_client_.find(key);
    if (it == request_client_.end()) return;           // client gone
    int cfd = it->second;
    if (terminal || ev == "listResult" || ev == "pong" || ev == "statsResult")
      request_client_.erase(it);
    if (!send_to_client(cfd, line)) drop_client(cfd);
  }

  void drop_client(int fd) {
    close(fd);
    clients_.erase(fd);
    for (auto it = request_client_.begin(); it != request_client_.end();) {
      if (it->second == fd) {
        const std::string& rid = it->first;
        if (rid != "list" && rid != "ping" && rid != "stats") {
          // stop the engine decoding for a dead client; tombstone the id so
          // late events are suppressed until the worker acks
          aborted_.insert(rid);
          write_all(worker_in_,
                    "{\"method\":\"abort\",\"requestId\":\"" + rid + "\"}\n");
        }
        it = request_client_.erase(it);
      } else {
        ++it;
      }
    }
  }

  void shutdown_all() {
    if (worker_pid_ > 0) {
      kill(worker_pid_, SIGTERM);
      int status = 0;
      waitpid(worker_pid_, &status, 0);
    }
    for (auto& [fd, c] : clients_) close(fd);
    if (listen_fd_ >= 0) close(listen_fd_);
    unlink(socket_path_.c_str());
  }

  void loop() {
    std::string pending;
    for (;;) {
      std::vector<pollfd> fds;
      fds.push_back({listen_fd_, POLLIN, 0});
      fds.push_back({worker_out_, POLLIN, 0});
      for (auto& [fd, c] : clients_)
        fds.push_back({fd, (short)(POLLIN | (c.outbuf.empty() ? 0 : POLLOUT)), 0});
      if (poll(fds.data(), fds.size(), 1000) < 0) {
        if (errno == EINTR) continue;
        break;
      }
      // worker died?  Error out every in-flight request (the reference
      // channel errors the request back to the caller — a respawned worker
      // knows nothing about old requestIds, so waiting clients would hang),
      // then respawn.
      int status;
      if (worker_pid_ > 0 && waitpid(worker_pid_, &status, WNOHANG) == worker_pid_) {
        fprintf(stderr, "[daemon] engine worker exited (%d); restarting\n", status);
        close(worker_in_);
        close(worker_out_);
        worker_pid_ = -1;
        worker_buf_.clear();
        std::map<std::string, int> inflight;
        inflight.swap(request_client_);
        aborted_.clear();  // old worker's ids can never arrive again
        for (auto& [rid, cfd] : inflight) {
          const char* ev = (rid == "list") ? "listResult" : (rid == "ping") ? "pong" : "onError";
          std::string msg = std::string("{\"event\":\"") + ev +
                            "\",\"requestId\":\"" + rid +
                            "\",\"message\":\"engine worker crashed; request lost\"}";
          if (!send_to_client(cfd, msg)) drop_client(cfd);
        }
        // crash-loop breaker: a worker that dies within 20 s of spawn
        // three times in a row will never come up (bad model, import
        // failure, OOM) — exit instead of forking torch forever.  Clients
        // see EOF; the launcher CLI reports the daemon down.
        if (time(nullptr) - last_spawn_ < 20) {
          if (++fast_crashes_ >= 3) {
            fprintf(stderr, "[daemon] worker crashed %d times at startup; giving up\n",
                    fast_crashes_);
            break;
          }
        } else {
          fast_crashes_ = 0;
        }
        if (!spawn_worker()) break;
        continue;
      }
      if (fds[0].revents & POLLIN) {
        int cfd = accept(listen_fd_, nullptr, nullptr);
        if (cfd >= 0) {
          int fl = fcntl(cfd, F_GETFL, 0);
          fcntl(cfd, F_SETFL, fl | O_NONBLOCK);
          clients_[cfd] = Client{cfd, "", ""};
        }
      }
      if (fds[1].revents & (POLLIN | POLLHUP)) {
        char buf[65536];
        ssize_t n = read(worker_out_, buf, sizeof(buf));
        if (n > 0) {
          worker_buf_.append(buf, (size_t)n);
          size_t pos;
          while ((pos = worker_buf_.find('\n')) != std::string::npos) {
            handle_worker_line(worker_buf_.substr(0, pos));
            worker_buf_.erase(0, pos + 1);
          }
        }
      }
      for (size_t i = 2; i < fds.size(); ++i) {
        int fd = fds[i].fd;
        if (fds[i].revents & POLLOUT) {
          auto it = clients_.find(fd);
          if (it != clients_.end() && !flush_client(it->second)) {
            drop_client(fd);
            continue;
          }
        }
        if (!(fds[i].revents & (POLLIN | POLLHUP | POLLERR))) continue;
        char buf[65536];
        ssize_t n = read(fd, buf, sizeof(buf));
        if (n <= 0) {
          if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) continue;
          drop_client(fd);
          continue;
        }
        auto& client = clients_[fd];
        client.inbuf.append(buf, (size_t)n);
        size_t pos;
        while ((pos = client.inbuf.find('\n')) != std::string::npos) {
          std::string line = client.inbuf.substr(0, pos);
          client.inbuf.erase(0, pos + 1);
          if (!line.empty()) handle_client_line(fd, line);
        }
      }
    }
    shutdown_all();
  }
};

}  // namespace

int main(int argc, char** argv) {
  std::string socket_path = "/tmp/senweaver_amd.sock";
  std::vector<std::string> worker{"python3", "-m", "senweaver_amd.server.engine_worker"};
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    if (a == "--socket" && i + 1 < argc) {
      socket_path = argv[++i];
    } else if (a == "--worker" && i + 1 < argc) {
      worker.clear();
      // split on spaces
      std::string w = argv[++i];
      size_t start = 0;
      while (start < w.size()) {
        size_t sp = w.find(' ', start);
        if (sp == std::string::npos) sp = w.size();
        if (sp > start) worker.push_back(w.substr(start, sp - start));
        start = sp + 1;
      }
    }
  }
  Daemon d(socket_path, worker);
  return d.run();
}
