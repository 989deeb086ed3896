// _frontend — native HTTP/1.1 ingestion front end for the MCP gateway.
//
// The reference serves each request on its own goroutine through Go's
// net/http (cmd/grmcp/main.go:202-208).  The asyncio surface here
// (server/http.py) mirrors that for capability parity, but tops out around
// ~170 us of interpreter work per request.  This module is the MI355X
// serving path: a C++ reactor accepts MCP POSTs, COLLECTS CONCURRENT
// REQUEST BODIES INTO BATCHES (adaptive window, exactly the shape
// k_json2pb wants), and hands each batch to Python in ONE GIL crossing
// (GpuPipeline.process_batch releases the GIL for the GPU/network stages).
// Responses return to the reactor over an eventfd and are written back on
// the right connections in arrival order.
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
#include <poll.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>
#include <string.h>
#include <sys/epoll.h>
#include <sys/eventfd.h>
#include <sys/socket.h>
#include <unistd.h>

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
#include <vector>

namespace py = pybind11;
using Clock = std::chrono::steady_clock;

namespace {

struct PendingReq {
  uint64_t conn_id;
  uint64_t seq;        // per-connection arrival order
  std::string body;
  std::string session; // Mcp-Session-Id or empty
  std::vector<std::pair<std::string, std::string>> headers;
  bool batchable;      // POST "/" with json content
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
};

std::string http_response(int status, const std::string& body,
                          const std::string& session_id) {
  const char* reason = status == 200 ? "OK"
                       : status == 429 ? "Too Many Requests"
                       : status == 404 ? "Not Found"
                       : status == 413 ? "Payload Too Large"
                       : status == 503 ? "Service Unavailable"
                                       : "Error";
  std::string out;
  out.reserve(body.size() + 256);
  out += "HTTP/1.1 " + std::to_string(status) + " " + reason + "\r\n";
  out += "Content-Type: application/json\r\n";
  // security headers (middleware.go:65-86)
  out += "X-Content-Type-Options: nosniff\r\nX-Frame-Options: DENY\r\n";
  if (!session_id.empty()) out += "Mcp-Session-Id: " + session_id + "\r\n";
  out += "Content-Length: " + std::to_string(body.size()) + "\r\n\r\n";
  out += body;
  return out;
}

bool iequal(const char* a, const char* b, size_t n) {
  for (size_t i = 0; i < n; ++i)
    if (tolower((unsigned char)a[i]) != tolower((unsigned char)b[i])) return false;
  return true;
}

}  // namespace

class Frontend {
 public:
  Frontend(const std::string& host, int port, py::function batch_cb,
           py::function slow_cb, int batch_window_us, int max_batch,
           size_t max_body, double rate_rps, double rate_burst,
           int workers)
      : host_(host), port_(port), batch_cb_(batch_cb), slow_cb_(slow_cb),
        window_us_(batch_window_us), max_batch_(max_batch),
        max_body_(max_body), rate_rps_(rate_rps),
        tokens_(rate_burst), burst_(rate_burst),
        n_workers_(workers < 1 ? 1 : workers) {}

  ~Frontend() { stop(); }

  int start() {
    listen_fd_ = socket(AF_INET, SOCK_STREAM, 0);
    int one = 1;
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons((uint16_t)port_);
    inet_pton(AF_INET, host_.c_str(), &addr.sin_addr);
    if (bind(listen_fd_, (sockaddr*)&addr, sizeof(addr)) != 0)
      throw std::runtime_error("frontend bind failed");
    socklen_t alen = sizeof(addr);
    getsockname(listen_fd_, (sockaddr*)&addr, &alen);
    port_ = ntohs(addr.sin_port);
    if (listen(listen_fd_, 1024) != 0) throw std::runtime_error("listen failed");
    fcntl(listen_fd_, F_SETFL, fcntl(listen_fd_, F_GETFL, 0) | O_NONBLOCK);

    epfd_ = epoll_create1(0);
    wake_fd_ = eventfd(0, EFD_NONBLOCK);
    epoll_event ev{};
    ev.events = EPOLLIN;
    ev.data.u64 = LISTEN_KEY;
    epoll_ctl(epfd_, EPOLL_CTL_ADD, listen_fd_, &ev);
    ev.data.u64 = WAKE_KEY;
    epoll_ctl(epfd_, EPOLL_CTL_ADD, wake_fd_, &ev);

    stop_.store(false);
    idle_workers_.store(n_workers_);
    io_thread_ = std::thread([this] { io_loop(); });
    for (int i = 0; i < n_workers_; ++i)
      worker_threads_.emplace_back([this] { worker_loop(); });
    return port_;
  }

  void stop() {
    if (stop_.exchange(true)) return;
    uint64_t one = 1;
    (void)!write(wake_fd_, &one, 8);
    {
      std::lock_guard<std::mutex> lk(batch_mu_);
      batch_cv_.notify_all();
    }
    if (io_thread_.joinable()) io_thread_.join();
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
    for (auto& kv : conns_) close(kv.second->fd);
    conns_.clear();
    if (listen_fd_ >= 0) close(listen_fd_);
    if (epfd_ >= 0) close(epfd_);
    if (wake_fd_ >= 0) close(wake_fd_);
    listen_fd_ = epfd_ = wake_fd_ = -1;
  }

  int port() const { return port_; }

 private:
  static constexpr uint64_t LISTEN_KEY = ~0ull;
  static constexpr uint64_t WAKE_KEY = ~0ull - 1;

  // ---- io reactor ---------------------------------------------------------

  void io_loop() {
    std::vector<epoll_event> events(256);
    while (!stop_.load()) {
      int timeout_ms = pending_.empty() ? 50 : 1;
      int n = epoll_wait(epfd_, events.data(), (int)events.size(), timeout_ms);
      auto now = Clock::now();
      for (int i = 0; i < n; ++i) {
        uint64_t key = events[i].data.u64;
        if (key == LISTEN_KEY) {
          accept_new();
        } else if (key == WAKE_KEY) {
          uint64_t junk;
          while (read(wake_fd_, &junk, 8) > 0) {}
          drain_completions();
        } else {
          handle_conn(key, events[i].events);
        }
      }
      drain_completions();
      // Backpressure-clocked batching: while the worker is busy with the
      // previous batch, arrivals accumulate; the moment it goes idle the
      // whole backlog ships as one batch.  Batch size self-tunes to the
      // service time (GPU + gRPC) with no artificial latency window —
      // window_us_ only caps the wait when the worker is idle and a
      // request just arrived (micro-coalescing across the same epoll wake).
      (void)now;
      if (!pending_.empty() &&
          (pending_.size() >= (size_t)max_batch_ ||
           idle_workers_.load(std::memory_order_acquire) > 0)) {
        std::lock_guard<std::mutex> lk(batch_mu_);
        batches_.emplace_back(std::move(pending_));
        pending_.clear();
        batch_cv_.notify_one();
      }
    }
  }

  void accept_new() {
    while (true) {
      int cfd = accept(listen_fd_, nullptr, nullptr);
      if (cfd < 0) return;
      fcntl(cfd, F_SETFL, fcntl(cfd, F_GETFL, 0) | O_NONBLOCK);
      int one = 1;
      setsockopt(cfd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      uint64_t id = next_conn_id_++;
      auto conn = std::make_unique<Conn>();
      conn->fd = cfd;
      epoll_event ev{};
      ev.events = EPOLLIN;
      ev.data.u64 = id;
      epoll_ctl(epfd_, EPOLL_CTL_ADD, cfd, &ev);
      conns_[id] = std::move(conn);
    }
  }

  void handle_conn(uint64_t id, uint32_t evmask) {
    auto it = conns_.find(id);
    if (it == conns_.end()) return;
    Conn* c = it->second.get();
    if (evmask & (EPOLLHUP | EPOLLERR)) {
      drop_conn(id);
      return;
    }
    if (evmask & EPOLLIN) {
      char buf[1 << 16];
      while (true) {
        ssize_t r = recv(c->fd, buf, sizeof(buf), 0);
        if (r > 0) {
          c->rbuf.append(buf, r);
          if (r < (ssize_t)sizeof(buf)) break;
        } else if (r == 0) {
          drop_conn(id);
          return;
        } else {
          if (errno == EAGAIN || errno == EWOULDBLOCK) break;
          drop_conn(id);
          return;
        }
      }
      parse_requests(id, c);
      if (conns_.find(id) == conns_.end()) return;  // dropped during parse
    }
    if (evmask & EPOLLOUT) flush_conn(id, c);
  }

  void parse_requests(uint64_t id, Conn* c) {
    while (true) {
      size_t hdr_end = c->rbuf.find("\r\n\r\n");
      if (hdr_end == std::string::npos) {
        if (c->rbuf.size() > 64 * 1024) drop_conn(id);  // oversized headers
        return;
      }
      // request line
      size_t line_end = c->rbuf.find("\r\n");
      std::string line = c->rbuf.substr(0, line_end);
      size_t sp1 = line.find(' ');
      size_t sp2 = line.find(' ', sp1 + 1);
      if (sp1 == std::string::npos || sp2 == std::string::npos) {
        drop_conn(id);
        return;
      }
      std::string method = line.substr(0, sp1);
      std::string path = line.substr(sp1 + 1, sp2 - sp1 - 1);
      // headers
      size_t clen = 0;
      std::string session;
      bool is_json = false;
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
        else if (name.size() == 12 && iequal(name.data(), "content-type", 12))
          is_json = value.find("application/json") != std::string::npos;
        else if (name.size() == 14 && iequal(name.data(), "mcp-session-id", 14))
          session = value;
        else
          hdrs.emplace_back(std::move(name), std::move(value));
        pos = eol + 2;
      }
      if (clen > max_body_) {
        enqueue_direct(c, http_response(413, "{\"error\":\"body too large\"}", ""));
        drop_after_flush(id, c);
        return;
      }
      size_t total = hdr_end + 4 + clen;
      if (c->rbuf.size() < total) return;  // need more bytes
      std::string body = c->rbuf.substr(hdr_end + 4, clen);
      c->rbuf.erase(0, total);

      uint64_t seq = c->next_seq++;
      if (!allow_rate()) {
        complete(id, seq, http_response(429, "{\"error\":\"rate limited\"}", session));
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
      if (!req.batchable && method == "POST" && path == "/") {
        complete(id, seq,
                 http_response(415, "{\"error\":\"content-type must be application/json\"}",
                               session));
        continue;
      }
      if (pending_.empty()) first_pending_ = Clock::now();
      pending_.push_back(std::move(req));
    }
  }

  bool allow_rate() {
    if (rate_rps_ <= 0) return true;
    auto now = Clock::now();
    double dt = std::chrono::duration<double>(now - last_refill_).count();
    last_refill_ = now;
    tokens_ = std::min(burst_, tokens_ + dt * rate_rps_);
    if (tokens_ < 1.0) return false;
    tokens_ -= 1.0;
    return true;
  }

  // queue a response for (conn, seq); write in order
  void complete(uint64_t id, uint64_t seq, std::string payload) {
    auto it = conns_.find(id);
    if (it == conns_.end()) return;
    Conn* c = it->second.get();
    c->ready[seq] = std::move(payload);
    while (true) {
      auto rit = c->ready.find(c->next_write);
      if (rit == c->ready.end()) break;
      c->wbuf += rit->second;
      c->ready.erase(rit);
      c->next_write++;
    }
    flush_conn(id, c);
  }

  void enqueue_direct(Conn* c, std::string payload) { c->wbuf += payload; }

  void flush_conn(uint64_t id, Conn* c) {
    while (!c->wbuf.empty()) {
      ssize_t w = send(c->fd, c->wbuf.data(), c->wbuf.size(), MSG_NOSIGNAL);
      if (w > 0) {
        c->wbuf.erase(0, (size_t)w);
      } else if (w < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
        epoll_event ev{};
        ev.events = EPOLLIN | EPOLLOUT;
        ev.data.u64 = id;
        epoll_ctl(epfd_, EPOLL_CTL_MOD, c->fd, &ev);
        return;
      } else {
        drop_conn(id);
        return;
      }
    }
    epoll_event ev{};
    ev.events = EPOLLIN;
    ev.data.u64 = id;
    epoll_ctl(epfd_, EPOLL_CTL_MOD, c->fd, &ev);
    if (c->closing) drop_conn(id);
  }

  void drop_after_flush(uint64_t id, Conn* c) {
    c->closing = true;
    flush_conn(id, c);
  }

  void drop_conn(uint64_t id) {
    auto it = conns_.find(id);
    if (it == conns_.end()) return;
    epoll_ctl(epfd_, EPOLL_CTL_DEL, it->second->fd, nullptr);
    close(it->second->fd);
    conns_.erase(it);
  }

  void drain_completions() {
    std::deque<OutResp> done;
    {
      std::lock_guard<std::mutex> lk(done_mu_);
      done.swap(done_);
    }
    for (auto& r : done) complete(r.conn_id, r.seq, std::move(r.payload));
  }

  // ---- worker: one GIL crossing per batch --------------------------------

  void worker_loop() {
    while (true) {
      std::vector<PendingReq> batch;
      {
        std::unique_lock<std::mutex> lk(batch_mu_);
        batch_cv_.wait(lk, [this] { return stop_.load() || !batches_.empty(); });
        if (stop_.load() && batches_.empty()) return;
        batch = std::move(batches_.front());
        // coalesce any batches queued while we slept
        batches_.pop_front();
        while (!batches_.empty() &&
               batch.size() + batches_.front().size() <= (size_t)max_batch_) {
          auto& nxt = batches_.front();
          batch.insert(batch.end(), std::make_move_iterator(nxt.begin()),
                       std::make_move_iterator(nxt.end()));
          batches_.pop_front();
        }
        idle_workers_.fetch_sub(1, std::memory_order_acq_rel);
      }
      std::vector<OutResp> out;
      out.reserve(batch.size());
      {
        py::gil_scoped_acquire gil;
        // split: batchable bodies -> batch_cb; the rest one-by-one -> slow_cb
        py::list bodies, sessions, headers;
        std::vector<size_t> batch_idx;
        for (size_t i = 0; i < batch.size(); ++i) {
          if (!batch[i].batchable) continue;
          batch_idx.push_back(i);
          bodies.append(py::bytes(batch[i].body));
          sessions.append(batch[i].session.empty()
                              ? py::object(py::none())
                              : py::object(py::str(batch[i].session)));
          py::dict h;
          for (auto& kv : batch[i].headers) h[py::str(kv.first)] = py::str(kv.second);
          headers.append(h);
        }
        if (py::len(bodies) > 0) {
          try {
            py::list res = batch_cb_(bodies, sessions, headers);
            for (size_t k = 0; k < batch_idx.size(); ++k) {
              py::tuple t = res[k].cast<py::tuple>();
              std::string body = t[0].cast<std::string>();
              std::string sid = t[1].cast<std::string>();
              out.push_back({batch[batch_idx[k]].conn_id, batch[batch_idx[k]].seq,
                             http_response(200, body, sid)});
            }
          } catch (const std::exception& e) {
            std::string err = std::string("{\"jsonrpc\":\"2.0\",\"id\":null,\"error\":"
                                          "{\"code\":-32603,\"message\":\"internal\"}}");
            for (size_t k = 0; k < batch_idx.size(); ++k)
              out.push_back({batch[batch_idx[k]].conn_id, batch[batch_idx[k]].seq,
                             http_response(200, err, "")});
          }
        }
        for (size_t i = 0; i < batch.size(); ++i) {
          if (batch[i].batchable) continue;
          int status = 200;
          std::string body, sid = batch[i].session;
          try {
            py::dict h;
            for (auto& kv : batch[i].headers) h[py::str(kv.first)] = py::str(kv.second);
            if (!sid.empty()) h["mcp-session-id"] = py::str(sid);
            py::tuple t = slow_cb_(py::str(batch[i].method), py::str(batch[i].path),
                                   h, py::bytes(batch[i].body))
                              .cast<py::tuple>();
            status = t[0].cast<int>();
            body = t[1].cast<std::string>();
            sid = t[2].cast<std::string>();
          } catch (const std::exception& e) {
            status = 500;
            body = "{\"error\":\"internal\"}";
          }
          out.push_back({batch[i].conn_id, batch[i].seq,
                         http_response(status, body, sid)});
        }
      }
      {
        std::lock_guard<std::mutex> lk(done_mu_);
        for (auto& r : out) done_.push_back(std::move(r));
      }
      idle_workers_.fetch_add(1, std::memory_order_acq_rel);
      uint64_t one = 1;
      (void)!write(wake_fd_, &one, 8);
    }
  }

  std::string host_;
  int port_;
  py::function batch_cb_, slow_cb_;
  int window_us_, max_batch_;
  size_t max_body_;
  double rate_rps_;
  double tokens_, burst_;
  Clock::time_point last_refill_ = Clock::now();

  int listen_fd_ = -1, epfd_ = -1, wake_fd_ = -1;
  std::atomic<bool> stop_{true};
  std::thread io_thread_;
  std::vector<std::thread> worker_threads_;
  int n_workers_ = 1;
  std::atomic<int> idle_workers_{0};
  uint64_t next_conn_id_ = 0;
  std::unordered_map<uint64_t, std::unique_ptr<Conn>> conns_;

  std::vector<PendingReq> pending_;
  Clock::time_point first_pending_;
  std::mutex batch_mu_;
  std::condition_variable batch_cv_;
  std::deque<std::vector<PendingReq>> batches_;
  std::mutex done_mu_;
  std::deque<OutResp> done_;
};

PYBIND11_MODULE(_frontend, m) {
  m.doc() = "native HTTP/1.1 batch ingestion front end for the MCP gateway";
  py::class_<Frontend>(m, "Frontend")
      .def(py::init<const std::string&, int, py::function, py::function, int,
                    int, size_t, double, double, int>(),
           py::arg("host"), py::arg("port"), py::arg("batch_cb"),
           py::arg("slow_cb"), py::arg("batch_window_us") = 200,
           py::arg("max_batch") = 4096, py::arg("max_body") = 1 << 20,
           py::arg("rate_rps") = 0.0, py::arg("rate_burst") = 0.0,
           py::arg("workers") = 1)
      .def("start", &Frontend::start)
      .def("stop", &Frontend::stop)
      .def_property_readonly("port", &Frontend::port);
}
