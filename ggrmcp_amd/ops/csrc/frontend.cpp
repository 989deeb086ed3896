// _frontend — native HTTP/1.1 ingestion front end for the MCP gateway.
//
// The reference serves each request on its own goroutine through Go's
// net/http (cmd/grmcp/main.go:202-208).  The asyncio surface here
// (server/http.py) mirrors that for capability parity, but tops out around
// ~170 us of interpreter work per request.  This module is the MI355X
// serving path: SHARDED C++ REACTORS accept MCP POSTs (a single reactor
// measured ~25 us/request of parse+epoll+write and capped serving at
// ~40k req/s), COLLECT CONCURRENT REQUEST BODIES INTO BATCHES
// (backpressure-clocked: while the workers are busy arrivals accumulate,
// so batch size self-tunes to the service time), and hand each batch to
// Python in ONE GIL crossing (GpuPipeline.process_batch releases the GIL
// for the GPU/network stages).  Completions return over per-reactor
// eventfds and are written back on the right connections in arrival order
// (HTTP/1.1 pipelining safe).
//
// Scope: HTTP/1.1 keep-alive, Content-Length bodies (no chunked), POST "/"
// on the batch path; everything else (GET /, /health, /metrics, OPTIONS)
// goes through the slow callback one request at a time — those endpoints
// are rare by construction.  Global token-bucket rate limit + body cap +
// security headers live here so the hot path never enters Python for
// rejected traffic (middleware.go:89-102, 164-178, 65-86).

#include <arpa/inet.h>
#include <errno.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/epoll.h>
#include <sys/eventfd.h>
#include <sys/socket.h>
#include <unistd.h>

#include "session_table.h"
#include "span_api.h"

#include <algorithm>
#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <memory>
#include <mutex>
#include <string>
#include <thread>
#include <unordered_map>
#include <unordered_set>
#include <vector>

namespace py = pybind11;
using Clock = std::chrono::steady_clock;

namespace {

struct PendingReq {
  uint64_t conn_id;
  uint64_t seq;        // per-connection arrival order
  std::string body;
  std::string session; // Mcp-Session-Id or empty; effective id after guard
  std::vector<std::pair<std::string, std::string>> headers;
  bool batchable;      // POST "/" with json content
  int verdict = 0;     // sesstab::V_* when the C++ session guard ran
  std::string method, path;
};

struct OutResp {
  uint64_t conn_id;
  uint64_t seq;
  std::string payload;  // full HTTP bytes
};

struct Conn {
  int fd = -1;
  std::string rbuf;
  std::string wbuf;
  uint64_t next_seq = 0;      // next request sequence to assign
  uint64_t next_write = 0;    // next response sequence to write
  std::unordered_map<uint64_t, std::string> ready;  // out-of-order responses
  bool closing = false;
  bool want_out = false;  // EPOLLOUT currently armed (skip redundant MODs)
};

std::string http_response(int status, const std::string& body,
                          const std::string& session_id, bool close = false) {
  const char* reason = status == 200 ? "OK"
                       : status == 429 ? "Too Many Requests"
                       : status == 404 ? "Not Found"
                       : status == 413 ? "Payload Too Large"
                       : status == 501 ? "Not Implemented"
                       : status == 503 ? "Service Unavailable"
                                       : "Error";
  std::string out;
  out.reserve(body.size() + 256);
  out += "HTTP/1.1 " + std::to_string(status) + " " + reason + "\r\n";
  out += "Content-Type: application/json\r\n";
  // security + CORS headers (middleware.go:65-86, 46-62); Mcp-Session-Id
  // must be exposed for browser MCP clients
  out += "X-Content-Type-Options: nosniff\r\nX-Frame-Options: DENY\r\n";
  out += "Access-Control-Allow-Origin: *\r\n";
  out += "Access-Control-Allow-Methods: GET, POST, OPTIONS\r\n";
  out += "Access-Control-Allow-Headers: Content-Type, Mcp-Session-Id, Authorization\r\n";
  out += "Access-Control-Expose-Headers: Mcp-Session-Id\r\n";
  if (!session_id.empty()) {
    // single sink for ids echoed into a header: strip anything that
    // could split the response (defense in depth behind the parse-time
    // sanitizer — Python callbacks also hand ids back through here)
    std::string sid;
    sid.reserve(session_id.size());
    for (unsigned char ch : session_id)
      if (ch >= 0x20 && ch != 0x7F) sid += (char)ch;
    if (!sid.empty()) out += "Mcp-Session-Id: " + sid + "\r\n";
  }
  if (close) out += "Connection: close\r\n";
  out += "Content-Length: " + std::to_string(body.size()) + "\r\n\r\n";
  out += body;
  return out;
}

// HTTP field values that flow into Python str / h2 metadata must be
// visible ASCII: a high/control byte would throw inside the worker's
// py::str conversion (process-fatal outside a catch) and is invalid
// gRPC metadata anyway.
bool printable_ascii(const std::string& v) {
  for (unsigned char c : v)
    if (c < 0x20 || c > 0x7E) return false;
  return true;
}

bool iequal(const char* a, const char* b, size_t n) {
  for (size_t i = 0; i < n; ++i)
    if (tolower((unsigned char)a[i]) != tolower((unsigned char)b[i])) return false;
  return true;
}

// conn_id layout: [reactor index : 8][serial : 56]
inline int reactor_of(uint64_t id) { return (int)(id >> 56); }

}  // namespace

class Frontend {
 public:
  Frontend(const std::string& host, int port, py::function batch_cb,
           py::function slow_cb, int batch_window_us, int max_batch,
           size_t max_body, double rate_rps, double rate_burst, int workers,
           int reactors)
      : host_(host), port_(port), batch_cb_(batch_cb), slow_cb_(slow_cb),
        window_us_(batch_window_us), max_batch_(max_batch),
        max_body_(max_body), rate_rps_(rate_rps), tokens_(rate_burst),
        burst_(rate_burst), n_workers_(workers < 1 ? 1 : workers),
        n_reactors_(reactors < 1 ? 1 : (reactors > 255 ? 255 : reactors)) {}

  ~Frontend() { stop(); }

  // Header forwarding filter (reference pkg/headers/filter.go semantics:
  // disabled -> none; blocked wins; forward_all still honors blocked; else
  // allow-list).  Applied at parse time so the hot batch callback receives
  // pre-filtered, LOWERCASED names (h2 requires lowercase on the wire).
  // Call before start(); only used for case-insensitive configs — the
  // Python fallback filter handles the case-sensitive variant.
  void set_header_filter(bool enabled, bool forward_all,
                         const std::vector<std::string>& allow,
                         const std::vector<std::string>& block) {
    auto lower = [](std::string s) {
      for (auto& ch : s) ch = (char)tolower((unsigned char)ch);
      return s;
    };
    hfilter_on_ = true;
    hfwd_enabled_ = enabled;
    hfwd_all_ = forward_all;
    hallow_.clear();
    hblock_.clear();
    for (auto& s : allow) hallow_.insert(lower(s));
    for (auto& s : block) hblock_.insert(lower(s));
  }

  // allow N gateway processes (one per GPU rank) to share ONE port: the
  // kernel load-balances accepted connections across listeners.  Call
  // before start().
  void set_reuse_port(bool on) { reuse_port_ = on; }

  // C++ session guard (pkg/session/manager.go semantics, see
  // session_table.h): the reactor resolves/creates the session and applies
  // blocked/rate-limit verdicts at parse time, so the batch callback never
  // runs per-request Python session code.  With a /dev/shm-backed table the
  // SAME state is shared by every serve_dp rank on the port.  Call before
  // start(); the table must outlive the frontend (pybind keep_alive).
  void set_session_table(sesstab::SessionTable* st, bool rate_limit_enabled) {
    sess_table_ = st;
    sess_rate_limit_ = rate_limit_enabled;
  }

  // Fully-native serving span (span_api.h): workers run the tools/call hot
  // path — GPU encode, gRPC invoke, GPU decode, response envelopes —
  // entirely in C++; fallback_cb is entered ONLY for the slots the span
  // reports (streaming / non-tools-call / host fallbacks / rejected
  // sessions).  fallback_cb(items) -> list[bytes], item = (kind, body,
  // session_id, headers, aux, tool_idx); kind -1 = blocked session,
  // -2 = rate-limited, else spanapi K_PY_*.  Executors and clients are
  // opaque handles (Engine.span_handle() / Client.raw_handle()); their
  // owners must outlive the frontend.  Call before start().
  void set_native_span(const std::vector<uintptr_t>& execs,
                       const std::vector<uintptr_t>& clients,
                       double timeout_s, int max_depth, int max_string,
                       long max_args, int enforce, int max_span_batch,
                       long max_span_bytes, py::function fallback_cb) {
    span_execs_.clear();
    span_mu_.clear();
    for (auto h : execs) {
      span_execs_.push_back((spanapi::ISpanExecutor*)h);
      span_mu_.push_back(std::make_unique<std::mutex>());
    }
    span_clients_.clear();
    for (auto h : clients) span_clients_.push_back((void*)h);
    span_timeout_s_ = timeout_s;
    span_max_depth_ = (uint32_t)max_depth;
    span_max_string_ = (uint32_t)max_string;
    span_max_args_ = (uint32_t)max_args;
    span_enforce_ = enforce;
    span_max_batch_ = max_span_batch > 0 ? max_span_batch : 4096;
    span_max_bytes_ = max_span_bytes > 0 ? (size_t)max_span_bytes : (8u << 20);
    fallback_cb_ = fallback_cb;
  }

  // per-stage counters of the native span path (for /metrics; mirrors the
  // Python EngineStats snapshot keys)
  py::dict native_stats() const {
    py::dict d;
    d["batches"] = ns_batches_.load();
    d["requests"] = ns_requests_.load();
    d["gpuOk"] = ns_final_.load();
    d["errors"] = ns_err_final_.load();
    d["hostFallbacks"] = ns_py_slots_.load();
    d["spanFailures"] = ns_span_fail_.load();
    d["encodeMs"] = ns_enc_us_.load() / 1e3;
    d["invokeMs"] = ns_inv_us_.load() / 1e3;
    d["decodeMs"] = ns_dec_us_.load() / 1e3;
    d["encodeGpuMs"] = ns_enc_gpu_us_.load() / 1e3;
    d["decodeGpuMs"] = ns_dec_gpu_us_.load() / 1e3;
    if (rprof_on_) {
      d["rprofBusyMs"] = rprof_.busy_ns.load() / 1e6;
      d["rprofRecvMs"] = rprof_.recv_ns.load() / 1e6;
      d["rprofParseMs"] = rprof_.parse_ns.load() / 1e6;
      d["rprofGuardMs"] = rprof_.guard_ns.load() / 1e6;
      d["rprofFlushMs"] = rprof_.flush_ns.load() / 1e6;
      d["rprofCycles"] = rprof_.cycles.load();
      d["rprofReqs"] = rprof_.reqs.load();
    }
    return d;
  }

  int start() {
    listen_fd_ = socket(AF_INET, SOCK_STREAM, 0);
    int one = 1;
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    if (reuse_port_)
      setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEPORT, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons((uint16_t)port_);
    inet_pton(AF_INET, host_.c_str(), &addr.sin_addr);
    if (bind(listen_fd_, (sockaddr*)&addr, sizeof(addr)) != 0)
      throw std::runtime_error("frontend bind failed");
    socklen_t alen = sizeof(addr);
    getsockname(listen_fd_, (sockaddr*)&addr, &alen);
    port_ = ntohs(addr.sin_port);
    // large backlog: closed-loop fleets (re)connect thousands of sessions
    // at once; a 1024 backlog made the kernel drop SYNs past it and the
    // 1 s retransmit idled those sessions (measured: 2048 sessions ran at
    // 63k req/s vs 305k with the deeper backlog).  The kernel clamps to
    // net.core.somaxconn.
    if (listen(listen_fd_, 65535) != 0)
      throw std::runtime_error("listen failed");
    fcntl(listen_fd_, F_SETFL, fcntl(listen_fd_, F_GETFL, 0) | O_NONBLOCK);

    stop_.store(false);
    draining_.store(false);
    inflight_.store(0);
    reactors_.clear();
    for (int r = 0; r < n_reactors_; ++r) {
      auto re = std::make_unique<Reactor>();
      re->index = r;
      re->epfd = epoll_create1(0);
      re->wake_fd = eventfd(0, EFD_NONBLOCK);
      epoll_event ev{};
      ev.events = EPOLLIN;
      ev.data.u64 = WAKE_KEY;
      epoll_ctl(re->epfd, EPOLL_CTL_ADD, re->wake_fd, &ev);
      if (r == 0) {
        ev.data.u64 = LISTEN_KEY;
        epoll_ctl(re->epfd, EPOLL_CTL_ADD, listen_fd_, &ev);
      }
      reactors_.push_back(std::move(re));
    }
    idle_workers_.store(n_workers_);
    for (auto& re : reactors_) {
      Reactor* raw = re.get();
      raw->thread = std::thread([this, raw] { io_loop(raw); });
    }
    for (int i = 0; i < n_workers_; ++i)
      worker_threads_.emplace_back([this] { worker_loop(); });
    return port_;
  }

  // Graceful drain (reference main.go:94-112's 30 s shutdown window, for
  // the native path): stop accepting, let queued + in-flight requests
  // finish, then a short grace for socket flushes.  Returns the number of
  // requests still unanswered at timeout (0 = clean drain).  Call before
  // stop().
  long drain(double timeout_s) {
    draining_.store(true, std::memory_order_relaxed);
    auto deadline =
        Clock::now() + std::chrono::duration_cast<Clock::duration>(
                           std::chrono::duration<double>(timeout_s));
    {
      // workers need the GIL for batch callbacks — release while waiting
      py::gil_scoped_release rel;
      while (inflight_.load(std::memory_order_acquire) > 0 &&
             Clock::now() < deadline)
        std::this_thread::sleep_for(std::chrono::milliseconds(5));
      std::this_thread::sleep_for(std::chrono::milliseconds(100));
    }
    return inflight_.load(std::memory_order_acquire);
  }

  void stop() {
    if (stop_.exchange(true)) return;
    for (auto& re : reactors_) {
      uint64_t one = 1;
      (void)!write(re->wake_fd, &one, 8);
    }
    {
      std::lock_guard<std::mutex> lk(batch_mu_);
      batch_cv_.notify_all();
    }
    for (auto& re : reactors_)
      if (re->thread.joinable()) re->thread.join();
    {
      // Workers may be blocked acquiring the GIL; if this thread holds it,
      // release while joining (stop() is reachable both from Python calls
      // that hold the GIL and from C++ teardown that doesn't).
      auto join_all = [this] {
        for (auto& t : worker_threads_)
          if (t.joinable()) t.join();
      };
      if (PyGILState_Check()) {
        py::gil_scoped_release rel;
        join_all();
      } else {
        join_all();
      }
      worker_threads_.clear();
    }
    for (auto& re : reactors_) {
      for (auto& kv : re->conns) close(kv.second->fd);
      re->conns.clear();
      if (re->epfd >= 0) close(re->epfd);
      if (re->wake_fd >= 0) close(re->wake_fd);
    }
    reactors_.clear();
    if (listen_fd_ >= 0) close(listen_fd_);
    listen_fd_ = -1;
  }

  int port() const { return port_; }

 private:
  static constexpr uint64_t LISTEN_KEY = ~0ull;
  static constexpr uint64_t WAKE_KEY = ~0ull - 1;

  struct Reactor {
    int index = 0;
    int epfd = -1;
    int wake_fd = -1;
    std::thread thread;
    uint64_t next_serial = 0;
    std::unordered_map<uint64_t, std::unique_ptr<Conn>> conns;
    std::vector<PendingReq> pending;
    Clock::time_point first_pending;
    std::mutex add_mu;
    std::vector<int> to_add;          // fds assigned by the acceptor
    std::mutex done_mu;
    std::deque<OutResp> done;         // completions routed back here
  };

  // ---- io reactors --------------------------------------------------------

  // GGRMCP_REACTOR_PROF=1: per-stage wall aggregates over all reactor
  // threads (exported via native_stats) — how busy the reactors are and
  // where the per-request microseconds go.  ~2 clock reads per epoll
  // cycle + 2 per request when on; zero reads when off.
  struct RProf {
    std::atomic<long> busy_ns{0}, recv_ns{0}, parse_ns{0}, guard_ns{0},
        flush_ns{0}, cycles{0}, reqs{0};
  };
  RProf rprof_;
  const bool rprof_on_ = getenv("GGRMCP_REACTOR_PROF") != nullptr;

  void io_loop(Reactor* re) {
    std::vector<epoll_event> events(256);
    while (!stop_.load()) {
      int timeout_ms = re->pending.empty() ? 50 : 1;
      int n = epoll_wait(re->epfd, events.data(), (int)events.size(), timeout_ms);
      Clock::time_point b0;
      if (rprof_on_) b0 = Clock::now();
      for (int i = 0; i < n; ++i) {
        uint64_t key = events[i].data.u64;
        if (key == LISTEN_KEY) {
          accept_new(re);
        } else if (key == WAKE_KEY) {
          uint64_t junk;
          while (read(re->wake_fd, &junk, 8) > 0) {}
          adopt_new(re);
          drain_completions(re);
        } else {
          handle_conn(re, key, events[i].events);
        }
      }
      drain_completions(re);
      if (rprof_on_) {
        rprof_.busy_ns.fetch_add(
            (long)std::chrono::duration_cast<std::chrono::nanoseconds>(
                Clock::now() - b0)
                .count(),
            std::memory_order_relaxed);
        rprof_.cycles.fetch_add(1, std::memory_order_relaxed);
      }
      // Backpressure-clocked batching: while all workers are busy,
      // arrivals accumulate; dispatch as soon as a worker is idle (or the
      // batch is full).  batch_window_us additionally caps how long a
      // request may sit in this reactor's pending list before it is queued
      // to the workers even with every worker busy — queued batches from
      // every reactor coalesce at worker pop, so sharding the reactors
      // doesn't shrink GPU batches.
      if (!re->pending.empty() &&
          (re->pending.size() >= (size_t)max_batch_ ||
           idle_workers_.load(std::memory_order_acquire) > 0 ||
           Clock::now() - re->first_pending >=
               std::chrono::microseconds(window_us_))) {
        std::lock_guard<std::mutex> lk(batch_mu_);
        batches_.emplace_back(std::move(re->pending));
        re->pending.clear();
        batch_cv_.notify_one();
      }
    }
  }

  void accept_new(Reactor* re0) {
    // draining: leave new connections in the backlog; they die with the
    // listener at stop()
    if (draining_.load(std::memory_order_relaxed)) return;
    while (true) {
      int cfd = accept(listen_fd_, nullptr, nullptr);
      if (cfd < 0) return;
      fcntl(cfd, F_SETFL, fcntl(cfd, F_GETFL, 0) | O_NONBLOCK);
      int one = 1;
      setsockopt(cfd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      // round-robin connections across reactors
      int target = (int)(rr_.fetch_add(1) % reactors_.size());
      Reactor* re = reactors_[target].get();
      if (re == re0) {
        adopt_fd(re, cfd);
      } else {
        {
          std::lock_guard<std::mutex> lk(re->add_mu);
          re->to_add.push_back(cfd);
        }
        uint64_t one64 = 1;
        (void)!write(re->wake_fd, &one64, 8);
      }
    }
  }

  void adopt_new(Reactor* re) {
    std::vector<int> fds;
    {
      std::lock_guard<std::mutex> lk(re->add_mu);
      fds.swap(re->to_add);
    }
    for (int fd : fds) adopt_fd(re, fd);
  }

  void adopt_fd(Reactor* re, int cfd) {
    uint64_t id =
        ((uint64_t)re->index << 56) | (re->next_serial++ & 0x00FFFFFFFFFFFFFFull);
    auto conn = std::make_unique<Conn>();
    conn->fd = cfd;
    epoll_event ev{};
    ev.events = EPOLLIN;
    ev.data.u64 = id;
    epoll_ctl(re->epfd, EPOLL_CTL_ADD, cfd, &ev);
    re->conns[id] = std::move(conn);
  }

  void handle_conn(Reactor* re, uint64_t id, uint32_t evmask) {
    auto it = re->conns.find(id);
    if (it == re->conns.end()) return;
    Conn* c = it->second.get();
    if (evmask & (EPOLLHUP | EPOLLERR)) {
      drop_conn(re, id);
      return;
    }
    if (evmask & EPOLLIN) {
      Clock::time_point t0;
      if (rprof_on_) t0 = Clock::now();
      char buf[1 << 16];
      while (true) {
        ssize_t r = recv(c->fd, buf, sizeof(buf), 0);
        if (r > 0) {
          c->rbuf.append(buf, r);
          if (r < (ssize_t)sizeof(buf)) break;
        } else if (r == 0) {
          drop_conn(re, id);
          return;
        } else {
          if (errno == EAGAIN || errno == EWOULDBLOCK) break;
          drop_conn(re, id);
          return;
        }
      }
      Clock::time_point t1;
      if (rprof_on_) {
        t1 = Clock::now();
        rprof_.recv_ns.fetch_add(
            (long)std::chrono::duration_cast<std::chrono::nanoseconds>(t1 - t0)
                .count(),
            std::memory_order_relaxed);
      }
      parse_requests(re, id, c);
      if (rprof_on_)
        rprof_.parse_ns.fetch_add(
            (long)std::chrono::duration_cast<std::chrono::nanoseconds>(
                Clock::now() - t1)
                .count(),
            std::memory_order_relaxed);
      if (re->conns.find(id) == re->conns.end()) return;  // dropped in parse
    }
    if (evmask & EPOLLOUT) flush_conn(re, id, c);
  }

  void parse_requests(Reactor* re, uint64_t id, Conn* c) {
    while (true) {
      size_t hdr_end = c->rbuf.find("\r\n\r\n");
      if (hdr_end == std::string::npos) {
        if (c->rbuf.size() > 64 * 1024) drop_conn(re, id);  // oversized headers
        return;
      }
      size_t line_end = c->rbuf.find("\r\n");
      std::string line = c->rbuf.substr(0, line_end);
      size_t sp1 = line.find(' ');
      size_t sp2 = line.find(' ', sp1 + 1);
      if (sp1 == std::string::npos || sp2 == std::string::npos) {
        drop_conn(re, id);
        return;
      }
      std::string method = line.substr(0, sp1);
      std::string path = line.substr(sp1 + 1, sp2 - sp1 - 1);
      size_t clen = 0;
      std::string session;
      bool is_json = false;
      bool has_te = false;
      std::vector<std::pair<std::string, std::string>> hdrs;
      size_t pos = line_end + 2;
      while (pos < hdr_end) {
        size_t eol = c->rbuf.find("\r\n", pos);
        size_t colon = c->rbuf.find(':', pos);
        if (colon == std::string::npos || colon > eol) break;
        std::string name = c->rbuf.substr(pos, colon - pos);
        size_t vstart = colon + 1;
        while (vstart < eol && c->rbuf[vstart] == ' ') ++vstart;
        std::string value = c->rbuf.substr(vstart, eol - vstart);
        if (name.size() == 14 && iequal(name.data(), "content-length", 14))
          clen = (size_t)strtoull(value.c_str(), nullptr, 10);
        else if (name.size() == 17 &&
                 iequal(name.data(), "transfer-encoding", 17))
          has_te = true;
        else if (name.size() == 12 && iequal(name.data(), "content-type", 12))
          is_json = value.find("application/json") != std::string::npos;
        else if (name.size() == 14 && iequal(name.data(), "mcp-session-id", 14))
          session = value;
        else if (!hfilter_on_) {
          hdrs.emplace_back(std::move(name), std::move(value));
        } else if (hfwd_enabled_) {
          for (auto& ch : name) ch = (char)tolower((unsigned char)ch);
          if (!hblock_.count(name) && (hfwd_all_ || hallow_.count(name)))
            hdrs.emplace_back(std::move(name), std::move(value));
        }
        pos = eol + 2;
      }
      // chunked (or any Transfer-Encoding) bodies are out of scope; a
      // body parsed as zero-length would desync the pipelined stream
      // (request-smuggling-style misparse — ADVICE r1), so refuse and
      // close instead of misreading the connection.
      if (has_te) {
        c->wbuf += http_response(
            501, "{\"error\":\"transfer-encoding not supported\"}", "", true);
        c->closing = true;
        flush_conn(re, id, c);
        return;
      }
      if (clen > max_body_) {
        c->wbuf += http_response(413, "{\"error\":\"body too large\"}", "");
        c->closing = true;
        flush_conn(re, id, c);
        return;
      }
      size_t total = hdr_end + 4 + clen;
      if (c->rbuf.size() < total) return;  // need more bytes
      std::string body = c->rbuf.substr(hdr_end + 4, clen);
      c->rbuf.erase(0, total);
      // sanitize values leaving the wire-parse layer (see printable_ascii)
      if (!session.empty() && !printable_ascii(session))
        session.clear();  // treated as absent -> fresh session id
      if (!hdrs.empty())
        hdrs.erase(std::remove_if(hdrs.begin(), hdrs.end(),
                                  [](const std::pair<std::string,
                                                     std::string>& kv) {
                                    return !printable_ascii(kv.first) ||
                                           !printable_ascii(kv.second);
                                  }),
                   hdrs.end());

      uint64_t seq = c->next_seq++;
      if (!allow_rate()) {
        complete(re, id, seq,
                 http_response(429, "{\"error\":\"rate limited\"}", session));
        continue;
      }
      PendingReq req;
      req.conn_id = id;
      req.seq = seq;
      req.body = std::move(body);
      req.session = session;
      req.headers = std::move(hdrs);
      req.method = method;
      req.path = path;
      req.batchable = (method == "POST" && path == "/" && is_json);
      if (rprof_on_) rprof_.reqs.fetch_add(1, std::memory_order_relaxed);
      if (req.batchable && sess_table_) {
        Clock::time_point g0;
        if (rprof_on_) g0 = Clock::now();
        bool created = false;
        std::string sid;
        req.verdict = sess_table_->guard(req.session.data(), req.session.size(),
                                         sess_rate_limit_, &sid, &created);
        req.session = std::move(sid);
        if (rprof_on_)
          rprof_.guard_ns.fetch_add(
              (long)std::chrono::duration_cast<std::chrono::nanoseconds>(
                  Clock::now() - g0)
                  .count(),
              std::memory_order_relaxed);
      }
      if (!req.batchable && method == "POST" && path == "/") {
        complete(re, id, seq,
                 http_response(
                     415, "{\"error\":\"content-type must be application/json\"}",
                     session));
        continue;
      }
      if (re->pending.empty()) re->first_pending = Clock::now();
      inflight_.fetch_add(1, std::memory_order_relaxed);
      re->pending.push_back(std::move(req));
    }
  }

  bool allow_rate() {
    if (rate_rps_ <= 0) return true;
    std::lock_guard<std::mutex> lk(rate_mu_);
    auto now = Clock::now();
    double dt = std::chrono::duration<double>(now - last_refill_).count();
    last_refill_ = now;
    tokens_ = std::min(burst_, tokens_ + dt * rate_rps_);
    if (tokens_ < 1.0) return false;
    tokens_ -= 1.0;
    return true;
  }

  // queue a response for (conn, seq); write in arrival order
  void complete(Reactor* re, uint64_t id, uint64_t seq, std::string payload) {
    auto it = re->conns.find(id);
    if (it == re->conns.end()) return;
    Conn* c = it->second.get();
    if (seq == c->next_write && c->ready.empty()) {
      // common case: in-order completion (closed-loop clients keep one
      // request in flight per connection) — skip the reorder map
      c->wbuf += payload;
      c->next_write++;
    } else {
      c->ready[seq] = std::move(payload);
      while (true) {
        auto rit = c->ready.find(c->next_write);
        if (rit == c->ready.end()) break;
        c->wbuf += rit->second;
        c->ready.erase(rit);
        c->next_write++;
      }
    }
    flush_conn(re, id, c);
  }

  void flush_conn(Reactor* re, uint64_t id, Conn* c) {
    while (!c->wbuf.empty()) {
      ssize_t w = send(c->fd, c->wbuf.data(), c->wbuf.size(), MSG_NOSIGNAL);
      if (w > 0) {
        c->wbuf.erase(0, (size_t)w);
      } else if (w < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
        if (!c->want_out) {
          epoll_event ev{};
          ev.events = EPOLLIN | EPOLLOUT;
          ev.data.u64 = id;
          epoll_ctl(re->epfd, EPOLL_CTL_MOD, c->fd, &ev);
          c->want_out = true;
        }
        return;
      } else {
        drop_conn(re, id);
        return;
      }
    }
    // only disarm EPOLLOUT when it was armed: the drained-buffer case is
    // the hot path (one MOD per response otherwise — measurable at 350k/s)
    if (c->want_out) {
      epoll_event ev{};
      ev.events = EPOLLIN;
      ev.data.u64 = id;
      epoll_ctl(re->epfd, EPOLL_CTL_MOD, c->fd, &ev);
      c->want_out = false;
    }
    if (c->closing) drop_conn(re, id);
  }

  void drop_conn(Reactor* re, uint64_t id) {
    auto it = re->conns.find(id);
    if (it == re->conns.end()) return;
    epoll_ctl(re->epfd, EPOLL_CTL_DEL, it->second->fd, nullptr);
    close(it->second->fd);
    re->conns.erase(it);
  }

  void drain_completions(Reactor* re) {
    std::deque<OutResp> done;
    {
      std::lock_guard<std::mutex> lk(re->done_mu);
      done.swap(re->done);
    }
    Clock::time_point f0;
    if (rprof_on_ && !done.empty()) f0 = Clock::now();
    for (auto& r : done) complete(re, r.conn_id, r.seq, std::move(r.payload));
    if (rprof_on_ && !done.empty())
      rprof_.flush_ns.fetch_add(
          (long)std::chrono::duration_cast<std::chrono::nanoseconds>(
              Clock::now() - f0)
              .count(),
          std::memory_order_relaxed);
    // decrement AFTER complete(): when drain() sees zero the responses are
    // already in connection write buffers (its grace period covers the
    // final socket flush)
    if (!done.empty())
      inflight_.fetch_sub((long)done.size(), std::memory_order_release);
  }

  // ---- workers: one GIL crossing per batch --------------------------------

  void worker_loop() {
    while (true) {
      std::vector<PendingReq> batch;
      {
        std::unique_lock<std::mutex> lk(batch_mu_);
        batch_cv_.wait(lk, [this] { return stop_.load() || !batches_.empty(); });
        if (stop_.load() && batches_.empty()) return;
        batch = std::move(batches_.front());
        batches_.pop_front();
        // coalesce batches queued (possibly from several reactors) while
        // the workers were busy.  GGRMCP_MERGE_MAX can cap the merge; the
        // measured default is UNCAPPED (= max_batch): capping at 512 cost
        // ~10% throughput for no p99 gain (profiles/mc_*.json — big
        // merges amortize the span fixed costs better than they straggle)
        static const size_t merge_env = [] {
          const char* e = getenv("GGRMCP_MERGE_MAX");
          size_t v = e ? (size_t)atoll(e) : (size_t)1 << 30;
          return v < 1 ? (size_t)1 : v;
        }();
        const size_t merge_cap =
            merge_env > (size_t)max_batch_ ? (size_t)max_batch_ : merge_env;
        while (!batches_.empty() &&
               batch.size() + batches_.front().size() <= merge_cap) {
          auto& nxt = batches_.front();
          batch.insert(batch.end(), std::make_move_iterator(nxt.begin()),
                       std::make_move_iterator(nxt.end()));
          batches_.pop_front();
        }
        idle_workers_.fetch_sub(1, std::memory_order_acq_rel);
      }
      std::vector<OutResp> out;
      out.reserve(batch.size());
      bool native = !span_execs_.empty();
      bool any_nonbatch = false;
      // slots the native span hands back to Python:
      // (batch index, kind, aux, tool_idx)
      std::vector<size_t> py_items;
      std::vector<int> py_kinds;
      std::vector<std::string> py_aux;
      std::vector<int32_t> py_tool;
      if (native) {
        // ---- GIL-FREE hot path: session verdicts already assigned by the
        // reactor; allowed tools/call slots run through the span executor
        // (GPU encode -> gRPC -> GPU decode -> envelope) in C++ ----------
        std::vector<size_t> ok_idx;
        for (size_t i = 0; i < batch.size(); ++i) {
          if (!batch[i].batchable) {
            any_nonbatch = true;
            continue;
          }
          if (batch[i].verdict != 0) {
            py_items.push_back(i);
            py_kinds.push_back(-batch[i].verdict);
            py_aux.emplace_back();
            py_tool.push_back(-1);
            continue;
          }
          ok_idx.push_back(i);
        }
        ns_batches_.fetch_add(ok_idx.empty() ? 0 : 1,
                              std::memory_order_relaxed);
        ns_requests_.fetch_add((long)ok_idx.size(), std::memory_order_relaxed);
        // chunk by count and staged bytes (engine arena caps)
        size_t base = 0;
        while (base < ok_idx.size()) {
          size_t end = base, bytes = 0;
          while (end < ok_idx.size() && (end - base) < (size_t)span_max_batch_ &&
                 (end == base ||
                  bytes + batch[ok_idx[end]].body.size() <= span_max_bytes_)) {
            bytes += batch[ok_idx[end]].body.size();
            ++end;
          }
          size_t cn = end - base;
          std::vector<const char*> bptr(cn);
          std::vector<size_t> blen(cn);
          std::vector<
              const std::vector<std::pair<std::string, std::string>>*>
              metas(cn);
          for (size_t k = 0; k < cn; ++k) {
            PendingReq& r = batch[ok_idx[base + k]];
            bptr[k] = r.body.data();
            blen[k] = r.body.size();
            metas[k] = r.headers.empty() ? nullptr : &r.headers;
          }
          spanapi::SpanIn sin{bptr.data(),
                              blen.data(),
                              cn,
                              metas.data(),
                              span_clients_.data(),
                              span_clients_.size(),
                              span_timeout_s_,
                              span_max_depth_,
                              span_max_string_,
                              span_max_args_,
                              span_enforce_};
          spanapi::SpanOut sout;
          std::string serr;
          size_t ei = span_rr_.fetch_add(1) % span_execs_.size();
          bool okrun;
          {
            std::lock_guard<std::mutex> lk(*span_mu_[ei]);
            okrun = span_execs_[ei]->run_span(sin, &sout, &serr);
            if (okrun) {
              // materialize responses while the executor is held: the blob
              // belongs to the engine and is reused by its next span
              for (size_t k = 0; k < cn; ++k) {
                spanapi::SlotOut& so = sout.slots[k];
                size_t i = ok_idx[base + k];
                if (so.kind == spanapi::K_FINAL) {
                  ns_final_.fetch_add(1, std::memory_order_relaxed);
                  out.push_back(
                      {batch[i].conn_id, batch[i].seq,
                       http_response(
                           200,
                           std::string((const char*)sout.blob + so.off,
                                       so.len),
                           batch[i].session)});
                } else if (so.kind == spanapi::K_ERR_FINAL) {
                  ns_err_final_.fetch_add(1, std::memory_order_relaxed);
                  out.push_back({batch[i].conn_id, batch[i].seq,
                                 http_response(200, so.aux,
                                               batch[i].session)});
                } else {
                  py_items.push_back(i);
                  py_kinds.push_back(so.kind);
                  py_aux.push_back(std::move(so.aux));
                  py_tool.push_back(so.tool_idx);
                }
              }
              ns_enc_us_.fetch_add((long)(sout.enc_ms * 1e3),
                                   std::memory_order_relaxed);
              ns_inv_us_.fetch_add((long)(sout.inv_ms * 1e3),
                                   std::memory_order_relaxed);
              ns_dec_us_.fetch_add((long)(sout.dec_ms * 1e3),
                                   std::memory_order_relaxed);
              ns_enc_gpu_us_.fetch_add((long)(sout.enc_gpu_ms * 1e3),
                                       std::memory_order_relaxed);
              ns_dec_gpu_us_.fetch_add((long)(sout.dec_gpu_ms * 1e3),
                                       std::memory_order_relaxed);
            }
          }
          if (!okrun) {
            // engine failure: answer -32603 rather than guessing whether
            // the invoke stage ran (a blind retry could double side
            // effects on non-idempotent methods)
            ns_span_fail_.fetch_add(1, std::memory_order_relaxed);
            std::string env =
                "{\"jsonrpc\":\"2.0\",\"id\":null,\"error\":{\"code\":-32603,"
                "\"message\":\"engine failure\"}}";
            for (size_t k = 0; k < cn; ++k) {
              size_t i = ok_idx[base + k];
              out.push_back({batch[i].conn_id, batch[i].seq,
                             http_response(200, env, batch[i].session)});
            }
          }
          base = end;
        }
        ns_py_slots_.fetch_add((long)py_items.size(),
                               std::memory_order_relaxed);
      } else {
        for (size_t i = 0; i < batch.size(); ++i)
          if (!batch[i].batchable) any_nonbatch = true;
      }
      bool need_python =
          any_nonbatch || !py_items.empty() || (!native && !batch.empty());
      if (need_python) {
        py::gil_scoped_acquire gil;
        if (!native) {
          py::list bodies, sessions, headers, verdicts;
          std::vector<size_t> batch_idx;
          bool guarded = sess_table_ != nullptr;
          try {
            for (size_t i = 0; i < batch.size(); ++i) {
              if (!batch[i].batchable) continue;
              batch_idx.push_back(i);
              bodies.append(py::bytes(batch[i].body));
              sessions.append(batch[i].session.empty()
                                  ? py::object(py::none())
                                  : py::object(py::str(batch[i].session)));
              if (guarded) verdicts.append(py::int_(batch[i].verdict));
              py::dict h;
              for (auto& kv : batch[i].headers)
                h[py::str(kv.first)] = py::str(kv.second);
              headers.append(h);
            }
          } catch (const std::exception&) {
            // conversion failure (should be impossible post-sanitize):
            // answer the whole batch with -32603 rather than dying
            std::string err =
                "{\"jsonrpc\":\"2.0\",\"id\":null,\"error\":"
                "{\"code\":-32603,\"message\":\"internal\"}}";
            for (size_t k = 0; k < batch_idx.size(); ++k)
              out.push_back({batch[batch_idx[k]].conn_id,
                             batch[batch_idx[k]].seq,
                             http_response(200, err, "")});
            batch_idx.clear();
            bodies = py::list();
          }
          if (py::len(bodies) > 0) {
            try {
              py::list res = batch_cb_(bodies, sessions, headers,
                                       guarded ? py::object(verdicts)
                                               : py::object(py::none()));
              for (size_t k = 0; k < batch_idx.size(); ++k) {
                py::tuple t = res[k].cast<py::tuple>();
                std::string body = t[0].cast<std::string>();
                std::string sid = t[1].cast<std::string>();
                out.push_back({batch[batch_idx[k]].conn_id,
                               batch[batch_idx[k]].seq,
                               http_response(200, body, sid)});
              }
            } catch (const std::exception& e) {
              std::string err = std::string(
                  "{\"jsonrpc\":\"2.0\",\"id\":null,\"error\":"
                  "{\"code\":-32603,\"message\":\"internal\"}}");
              for (size_t k = 0; k < batch_idx.size(); ++k)
                out.push_back({batch[batch_idx[k]].conn_id,
                               batch[batch_idx[k]].seq,
                               http_response(200, err, "")});
            }
          }
        } else if (!py_items.empty()) {
          // rare slots the span could not finish natively
          try {
            py::list items;
            for (size_t k = 0; k < py_items.size(); ++k) {
              PendingReq& r = batch[py_items[k]];
              py::dict h;
              for (auto& kv : r.headers)
                h[py::str(kv.first)] = py::str(kv.second);
              items.append(py::make_tuple(
                  py_kinds[k], py::bytes(r.body), py::str(r.session), h,
                  py::bytes(py_aux[k]), py_tool[k]));
            }
            py::list res = fallback_cb_(items);
            for (size_t k = 0; k < py_items.size(); ++k) {
              PendingReq& r = batch[py_items[k]];
              out.push_back({r.conn_id, r.seq,
                             http_response(200, res[k].cast<std::string>(),
                                           r.session)});
            }
          } catch (const std::exception& e) {
            std::string env =
                "{\"jsonrpc\":\"2.0\",\"id\":null,\"error\":{\"code\":-32603,"
                "\"message\":\"internal\"}}";
            for (size_t k = 0; k < py_items.size(); ++k) {
              PendingReq& r = batch[py_items[k]];
              out.push_back(
                  {r.conn_id, r.seq, http_response(200, env, r.session)});
            }
          }
        }
        for (size_t i = 0; i < batch.size(); ++i) {
          if (batch[i].batchable) continue;
          int status = 200;
          std::string body, sid = batch[i].session;
          try {
            py::dict h;
            for (auto& kv : batch[i].headers)
              h[py::str(kv.first)] = py::str(kv.second);
            if (!sid.empty()) h["mcp-session-id"] = py::str(sid);
            py::tuple t =
                slow_cb_(py::str(batch[i].method), py::str(batch[i].path), h,
                         py::bytes(batch[i].body))
                    .cast<py::tuple>();
            status = t[0].cast<int>();
            body = t[1].cast<std::string>();
            sid = t[2].cast<std::string>();
          } catch (const std::exception& e) {
            status = 500;
            body = "{\"error\":\"internal\"}";
          }
          out.push_back(
              {batch[i].conn_id, batch[i].seq, http_response(status, body, sid)});
        }
      }
      // route completions to their reactors
      for (auto& r : out) {
        int ri = reactor_of(r.conn_id);
        if (ri < 0 || ri >= (int)reactors_.size()) continue;
        Reactor* re = reactors_[ri].get();
        std::lock_guard<std::mutex> lk(re->done_mu);
        re->done.push_back(std::move(r));
      }
      idle_workers_.fetch_add(1, std::memory_order_acq_rel);
      for (auto& re : reactors_) {
        uint64_t one = 1;
        (void)!write(re->wake_fd, &one, 8);
      }
    }
  }

  std::string host_;
  int port_;
  py::function batch_cb_, slow_cb_;
  // native span state (set_native_span)
  std::vector<spanapi::ISpanExecutor*> span_execs_;
  std::vector<std::unique_ptr<std::mutex>> span_mu_;
  std::vector<void*> span_clients_;
  double span_timeout_s_ = 30.0;
  uint32_t span_max_depth_ = 10, span_max_string_ = 1024,
           span_max_args_ = 1u << 20;
  int span_enforce_ = 1;
  int span_max_batch_ = 4096;
  size_t span_max_bytes_ = 8u << 20;
  py::function fallback_cb_;
  std::atomic<size_t> span_rr_{0};
  std::atomic<long> ns_batches_{0}, ns_requests_{0}, ns_final_{0},
      ns_err_final_{0}, ns_py_slots_{0}, ns_span_fail_{0};
  std::atomic<long> ns_enc_us_{0}, ns_inv_us_{0}, ns_dec_us_{0},
      ns_enc_gpu_us_{0}, ns_dec_gpu_us_{0};
  bool reuse_port_ = false;
  sesstab::SessionTable* sess_table_ = nullptr;
  bool sess_rate_limit_ = true;
  bool hfilter_on_ = false, hfwd_enabled_ = true, hfwd_all_ = false;
  std::unordered_set<std::string> hallow_, hblock_;
  int window_us_, max_batch_;
  size_t max_body_;
  double rate_rps_;
  double tokens_, burst_;
  std::mutex rate_mu_;
  Clock::time_point last_refill_ = Clock::now();

  int listen_fd_ = -1;
  std::atomic<bool> stop_{true};
  std::atomic<bool> draining_{false};
  std::atomic<long> inflight_{0};
  std::atomic<uint64_t> rr_{0};
  std::vector<std::unique_ptr<Reactor>> reactors_;
  std::vector<std::thread> worker_threads_;
  int n_workers_ = 1;
  int n_reactors_ = 1;
  std::atomic<int> idle_workers_{0};

  std::mutex batch_mu_;
  std::condition_variable batch_cv_;
  std::deque<std::vector<PendingReq>> batches_;
};

// ---------------------------------------------------------------------------
// MockSpanExecutor — CPU stand-in for the GPU engine so the frontend's
// native-span plumbing (chunking, per-kind fallback routing, blob copies,
// error paths) is testable without a GPU.  Body markers select the slot
// kind; anything else echoes a canned K_FINAL response.
// ---------------------------------------------------------------------------

class MockSpanExecutor : public spanapi::ISpanExecutor {
 public:
  bool run_span(const spanapi::SpanIn& in, spanapi::SpanOut* out,
                std::string* err) override {
    using namespace spanapi;
    calls_++;
    if (fail_next_) {
      fail_next_ = false;
      *err = "mock failure";
      return false;
    }
    blob_.clear();
    out->slots.assign(in.n, SlotOut());
    for (size_t i = 0; i < in.n; ++i) {
      std::string body(in.bodies[i], in.body_lens[i]);
      SlotOut& so = out->slots[i];
      so.tool_idx = (int32_t)i;
      if (body.find("__stream__") != std::string::npos) {
        so.kind = K_PY_STREAM;
        so.aux = "PB";
      } else if (body.find("__notool__") != std::string::npos) {
        so.kind = K_PY_NOT_TOOLCALL;
      } else if (body.find("__encfb__") != std::string::npos) {
        so.kind = K_PY_ENC_FALLBACK;
      } else if (body.find("__decfb__") != std::string::npos) {
        so.kind = K_PY_DEC_FALLBACK;
        so.aux = "WIRE";
      } else if (body.find("__err__") != std::string::npos) {
        so.kind = K_ERR_FINAL;
        so.aux =
            "{\"jsonrpc\":\"2.0\",\"id\":null,\"error\":{\"code\":-32600,"
            "\"message\":\"mock\"}}";
      } else {
        so.kind = K_FINAL;
        so.off = (uint32_t)blob_.size();
        std::string resp = "{\"jsonrpc\":\"2.0\",\"id\":1,\"result\":{"
                           "\"content\":[{\"type\":\"text\",\"text\":\"n=" +
                           std::to_string(in.body_lens[i]) +
                           "\"}],\"isError\":false}}";
        blob_ += resp;
        so.len = (uint32_t)resp.size();
      }
    }
    out->blob = (const uint8_t*)blob_.data();
    out->blob_len = blob_.size();
    out->enc_ms = 0.01;
    out->inv_ms = 0.01;
    out->dec_ms = 0.01;
    return true;
  }
  void fail_next() { fail_next_ = true; }
  long calls() const { return calls_; }

 private:
  std::string blob_;
  bool fail_next_ = false;
  long calls_ = 0;
};

// ---------------------------------------------------------------------------
// load generator — C++ closed-loop HTTP client for measuring the gateway
// without a Python client in the way.  N sessions across T threads, each
// session a keep-alive connection issuing sequential POSTs; returns
// (total_requests, elapsed_s, [p50_us, p90_us, p99_us], errors).
// ---------------------------------------------------------------------------

static py::tuple bench_client(const std::string& host, int port, int sessions,
                              int requests, const std::string& body,
                              int threads) {
  if (threads < 1) threads = 1;
  if (sessions < threads) threads = sessions;
  std::string req_head =
      "POST / HTTP/1.1\r\nHost: b\r\nContent-Type: application/json\r\n";
  std::atomic<long> errors{0};
  std::vector<std::vector<uint32_t>> lat(threads);
  auto run = [&](int t) {
    int per = sessions / threads + (t < sessions % threads ? 1 : 0);
    std::vector<int> fds;
    for (int s = 0; s < per; ++s) {
      int fd = socket(AF_INET, SOCK_STREAM, 0);
      sockaddr_in addr{};
      addr.sin_family = AF_INET;
      addr.sin_port = htons((uint16_t)port);
      inet_pton(AF_INET, host.c_str(), &addr.sin_addr);
      if (connect(fd, (sockaddr*)&addr, sizeof(addr)) != 0) {
        close(fd);
        errors++;
        continue;
      }
      int one = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      fds.push_back(fd);
    }
    // closed-loop: each connection keeps exactly one request in flight;
    // drive them round-robin with poll-free sequential turns per epoch
    std::vector<std::string> reqs(fds.size());
    for (size_t s = 0; s < fds.size(); ++s) {
      std::string sid = "bench-" + std::to_string(t) + "-" + std::to_string(s);
      reqs[s] = req_head + "Mcp-Session-Id: " + sid +
                "\r\nContent-Length: " + std::to_string(body.size()) +
                "\r\n\r\n" + body;
    }
    std::vector<Clock::time_point> t0(fds.size());
    lat[t].reserve(fds.size() * requests);
    // epoll-driven: send all, then read completions and resend
    int ep = epoll_create1(0);
    std::vector<int> remaining(fds.size(), requests);
    std::vector<std::string> bufs(fds.size());
    for (size_t s = 0; s < fds.size(); ++s) {
      epoll_event ev{};
      ev.events = EPOLLIN;
      ev.data.u64 = s;
      epoll_ctl(ep, EPOLL_CTL_ADD, fds[s], &ev);
      t0[s] = Clock::now();
      if (send(fds[s], reqs[s].data(), reqs[s].size(), MSG_NOSIGNAL) < 0) errors++;
    }
    size_t live = fds.size();
    std::vector<epoll_event> events(64);
    auto deadline = Clock::now() + std::chrono::seconds(120);
    while (live > 0 && Clock::now() < deadline) {
      int n = epoll_wait(ep, events.data(), (int)events.size(), 1000);
      for (int i = 0; i < n; ++i) {
        size_t s = events[i].data.u64;
        char rb[1 << 16];
        ssize_t r = recv(fds[s], rb, sizeof(rb), 0);
        if (r <= 0) {
          if (r < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) continue;
          errors++;
          epoll_ctl(ep, EPOLL_CTL_DEL, fds[s], nullptr);
          close(fds[s]);
          live--;
          continue;
        }
        bufs[s].append(rb, r);
        // complete iff we can see the whole response (headers+body)
        size_t he = bufs[s].find("\r\n\r\n");
        if (he == std::string::npos) continue;
        size_t cl = 0;
        size_t p = bufs[s].find("Content-Length:");
        if (p != std::string::npos && p < he)
          cl = (size_t)strtoull(bufs[s].c_str() + p + 15, nullptr, 10);
        if (bufs[s].size() < he + 4 + cl) continue;
        bufs[s].erase(0, he + 4 + cl);
        lat[t].push_back((uint32_t)std::chrono::duration_cast<std::chrono::microseconds>(
            Clock::now() - t0[s]).count());
        if (--remaining[s] <= 0) {
          epoll_ctl(ep, EPOLL_CTL_DEL, fds[s], nullptr);
          close(fds[s]);
          live--;
          continue;
        }
        t0[s] = Clock::now();
        if (send(fds[s], reqs[s].data(), reqs[s].size(), MSG_NOSIGNAL) < 0) {
          errors++;
          epoll_ctl(ep, EPOLL_CTL_DEL, fds[s], nullptr);
          close(fds[s]);
          live--;
        }
      }
    }
    close(ep);
  };
  auto t_start = Clock::now();
  std::vector<std::thread> ts;
  {
    py::gil_scoped_release rel;
    for (int t = 0; t < threads; ++t) ts.emplace_back(run, t);
    for (auto& th : ts) th.join();
  }
  double elapsed = std::chrono::duration<double>(Clock::now() - t_start).count();
  std::vector<uint32_t> all;
  for (auto& v : lat) all.insert(all.end(), v.begin(), v.end());
  std::sort(all.begin(), all.end());
  py::list pct;
  if (!all.empty()) {
    pct.append(all[all.size() / 2]);
    pct.append(all[(size_t)(all.size() * 0.9)]);
    pct.append(all[(size_t)(all.size() * 0.99)]);
  }
  return py::make_tuple((long)all.size(), elapsed, pct, errors.load());
}

PYBIND11_MODULE(_frontend, m) {
  m.doc() = "native HTTP/1.1 batch ingestion front end for the MCP gateway";
  py::class_<sesstab::SessionTable>(m, "SessionTable")
      .def(py::init<uint64_t, double, const std::string&, uint32_t, uint32_t>(),
           py::arg("capacity") = 16384, py::arg("ttl_s") = 1800.0,
           py::arg("path") = std::string(), py::arg("rate_per_min") = 100,
           py::arg("rate_burst") = 20,
           "Shared-memory session table (manager.go semantics).  path='' -> "
           "anonymous (single process); a /dev/shm path is shared by every "
           "rank that opens it (serve_dp session affinity).")
      .def("guard",
           [](sesstab::SessionTable& t, const std::string& id, bool rate_limit) {
             std::string out;
             bool created;
             int v = t.guard(id.data(), id.size(), rate_limit, &out, &created);
             return py::make_tuple(py::str(out), v, created);
           },
           py::arg("session_id") = std::string(), py::arg("rate_limit") = true,
           "-> (effective_id, verdict 0=ok/1=blocked/2=rate-limited, created)")
      .def("get_or_create",
           [](sesstab::SessionTable& t, const std::string& id) {
             return t.get_or_create(id.data(), id.size());
           },
           py::arg("session_id") = std::string())
      .def("block",
           [](sesstab::SessionTable& t, const std::string& id) {
             return t.set_blocked(id.data(), id.size(), true);
           })
      .def("unblock",
           [](sesstab::SessionTable& t, const std::string& id) {
             return t.set_blocked(id.data(), id.size(), false);
           })
      .def("remove",
           [](sesstab::SessionTable& t, const std::string& id) {
             return t.remove(id.data(), id.size());
           })
      .def("info",
           [](sesstab::SessionTable& t, const std::string& id) -> py::object {
             int64_t created_us, last_us;
             uint64_t calls;
             bool blocked;
             if (!t.info(id.data(), id.size(), &created_us, &last_us, &calls,
                         &blocked))
               return py::none();
             py::dict d;
             d["id"] = id;
             d["createdAt"] = created_us / 1e6;
             d["lastAccessed"] = last_us / 1e6;
             d["callCount"] = (long long)calls;
             d["isBlocked"] = blocked;
             return d;
           })
      .def("stats",
           [](sesstab::SessionTable& t) {
             uint64_t active, calls, blocked, created;
             t.stats(&active, &calls, &blocked, &created);
             py::dict d;
             d["activeSessions"] = (long long)active;
             d["totalCalls"] = (long long)calls;
             d["blockedSessions"] = (long long)blocked;
             d["createdTotal"] = (long long)created;
             d["maxSessions"] = (long long)t.capacity();
             return d;
           })
      .def_property_readonly("capacity", &sesstab::SessionTable::capacity);
  py::class_<Frontend>(m, "Frontend")
      .def(py::init<const std::string&, int, py::function, py::function, int,
                    int, size_t, double, double, int, int>(),
           py::arg("host"), py::arg("port"), py::arg("batch_cb"),
           py::arg("slow_cb"), py::arg("batch_window_us") = 200,
           py::arg("max_batch") = 4096, py::arg("max_body") = 1 << 20,
           py::arg("rate_rps") = 0.0, py::arg("rate_burst") = 0.0,
           py::arg("workers") = 1, py::arg("reactors") = 4)
      .def("drain", &Frontend::drain, py::arg("timeout_s"),
           "graceful drain: stop accepting, finish in-flight requests; "
           "returns requests still unanswered at timeout")
      .def("set_reuse_port", &Frontend::set_reuse_port, py::arg("on"),
           "SO_REUSEPORT: N gateway ranks share one port (call before start)")
      .def("set_session_table", &Frontend::set_session_table,
           py::arg("table"), py::arg("rate_limit_enabled") = true,
           py::keep_alive<1, 2>(),
           "run the per-request session guard in the C++ reactor "
           "(call before start)")
      .def("set_header_filter", &Frontend::set_header_filter,
           py::arg("enabled"), py::arg("forward_all"), py::arg("allow"),
           py::arg("block"),
           "install the forwarding filter in the parser (call before start)")
      .def("set_native_span", &Frontend::set_native_span, py::arg("execs"),
           py::arg("clients"), py::arg("timeout_s") = 30.0,
           py::arg("max_depth") = 10, py::arg("max_string") = 1024,
           py::arg("max_args") = (long)(1u << 20), py::arg("enforce") = 1,
           py::arg("max_span_batch") = 4096,
           py::arg("max_span_bytes") = (long)(8u << 20),
           py::arg("fallback_cb"),
           "run tools/call batches through ISpanExecutor handles in C++ "
           "(no GIL on the hot path); call before start")
      .def("native_stats", &Frontend::native_stats)
      .def("start", &Frontend::start)
      .def("stop", &Frontend::stop)
      .def_property_readonly("port", &Frontend::port);
  m.def("bench_client", &bench_client, py::arg("host"), py::arg("port"),
        py::arg("sessions"), py::arg("requests"), py::arg("body"),
        py::arg("threads") = 8);
  py::class_<MockSpanExecutor>(m, "MockSpanExecutor")
      .def(py::init<>())
      .def("span_handle",
           [](MockSpanExecutor& e) {
             return (uintptr_t)(spanapi::ISpanExecutor*)&e;
           })
      .def("fail_next", &MockSpanExecutor::fail_next)
      .def("calls", &MockSpanExecutor::calls);
}
