#pragma once
// h2grpc_impl.h — native gRPC-over-HTTP/2 transport (client + bench
// server) implementation.  Included by BOTH the thin _h2grpc module
// wrapper (h2grpc.cpp) and the engine module (engine.cpp), which embeds
// the client for the fully-native span executor.
//
// The gateway's hot path is: k_json2pb (GPU) -> gRPC unary invoke (host I/O)
// -> k_pb2json (GPU).  The reference does the invoke through Go's grpc stack
// (aalobaidi/ggRMCP pkg/grpc/reflection.go:367-373); a Python grpcio stub
// costs ~150 us of interpreter/C-core overhead per call and caps the whole
// gateway at a few k req/s.  This module replaces the I/O stage with a thin
// C++ client over nghttp2 (h2c prior knowledge, HPACK included): N
// connections, one event-loop thread each, thousands of multiplexed streams,
// batch submission with the GIL released.
//
// Also provides H2Server — a native backend implementing the bench services
// (hello.HelloService/SayHello parses its request and builds a real
// response; echo routes length-walk-validate the protobuf payload before
// echoing) so gateway measurements aren't bounded by a Python backend.
//
// gRPC wire framing: each HTTP/2 DATA payload carries 5-byte prefixed
// messages (compressed flag + u32 BE length).  Unary calls send exactly one.

#include <arpa/inet.h>
#include <errno.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <nghttp2/nghttp2.h>
#include <poll.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>
#include <string.h>
#include <stdio.h>
#include <stdlib.h>
#include <sys/eventfd.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <memory>
#include <mutex>
#include <stdexcept>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

namespace py = pybind11;
using Clock = std::chrono::steady_clock;

// ---------------------------------------------------------------------------
// helpers
// ---------------------------------------------------------------------------

// socket buffer size for the batch transport (both directions move whole
// 16 MB+ chunk batches; bigger kernel buffers mean fewer syscall round
// trips).  GGRMCP_SOCKBUF overrides; 0 keeps the kernel default.
static int transport_sockbuf() {
  const char* e = getenv("GGRMCP_SOCKBUF");
  return e ? atoi(e) : (4 << 20);
}

static void apply_sockbuf(int fd) {
  int sz = transport_sockbuf();
  if (sz > 0) {
    setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &sz, sizeof(sz));
    setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &sz, sizeof(sz));
  }
}

static int connect_target(const std::string& target) {
  int fd = -1;
  if (target.rfind("unix:", 0) == 0) {
    fd = socket(AF_UNIX, SOCK_STREAM, 0);
    if (fd < 0) throw std::runtime_error("socket() failed");
    sockaddr_un addr{};
    addr.sun_family = AF_UNIX;
    std::string path = target.substr(5);
    if (path.size() >= sizeof(addr.sun_path)) throw std::runtime_error("uds path too long");
    memcpy(addr.sun_path, path.c_str(), path.size() + 1);
    apply_sockbuf(fd);
    if (connect(fd, (sockaddr*)&addr, sizeof(addr)) != 0) {
      close(fd);
      throw std::runtime_error("connect failed: " + target);
    }
  } else {
    auto colon = target.rfind(':');
    if (colon == std::string::npos) throw std::runtime_error("bad target " + target);
    std::string host = target.substr(0, colon);
    int port = std::stoi(target.substr(colon + 1));
    fd = socket(AF_INET, SOCK_STREAM, 0);
    if (fd < 0) throw std::runtime_error("socket() failed");
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons((uint16_t)port);
    if (inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1) {
      close(fd);
      throw std::runtime_error("bad host (use a literal IP): " + host);
    }
    apply_sockbuf(fd);
    if (connect(fd, (sockaddr*)&addr, sizeof(addr)) != 0) {
      close(fd);
      throw std::runtime_error("connect failed: " + target);
    }
    int one = 1;
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
  }
  int flags = fcntl(fd, F_GETFL, 0);
  fcntl(fd, F_SETFL, flags | O_NONBLOCK);
  return fd;
}

#define NV(NAME, VALUE, NLEN, VLEN)                                      \
  nghttp2_nv {                                                           \
    (uint8_t*)(NAME), (uint8_t*)(VALUE), (size_t)(NLEN), (size_t)(VLEN), \
        NGHTTP2_NV_FLAG_NONE                                             \
  }

static nghttp2_nv nv(const char* name, const char* value) {
  return NV(name, value, strlen(name), strlen(value));
}
static nghttp2_nv nv(const char* name, const std::string& value) {
  return NV(name, value.data(), strlen(name), value.size());
}
static nghttp2_nv nv(const std::string& name, const std::string& value) {
  return NV(name.data(), value.data(), name.size(), value.size());
}

// grpc message framing
static std::string grpc_frame(const std::string& payload) {
  std::string out;
  out.resize(5 + payload.size());
  out[0] = 0;
  uint32_t n = htonl((uint32_t)payload.size());
  memcpy(&out[1], &n, 4);
  memcpy(&out[5], payload.data(), payload.size());
  return out;
}

// protobuf wire walk: validates structure, returns false on malformed input
static bool pb_validate(const uint8_t* p, size_t len) {
  size_t pos = 0;
  int depth_guard = 0;
  while (pos < len) {
    uint64_t tag = 0;
    int shift = 0;
    bool done = false;
    while (pos < len && shift <= 63) {
      uint8_t b = p[pos++];
      tag |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) { done = true; break; }
      shift += 7;
    }
    if (!done) return false;
    uint32_t wt = tag & 7;
    if ((tag >> 3) == 0) return false;
    switch (wt) {
      case 0: {  // varint
        bool vdone = false;
        for (int i = 0; i < 10 && pos < len; ++i) {
          if (!(p[pos++] & 0x80)) { vdone = true; break; }
        }
        if (!vdone) return false;
        break;
      }
      case 1: if (pos + 8 > len) return false; pos += 8; break;
      case 5: if (pos + 4 > len) return false; pos += 4; break;
      case 2: {
        uint64_t l = 0;
        int s2 = 0;
        bool d2 = false;
        while (pos < len && s2 <= 35) {
          uint8_t b = p[pos++];
          l |= (uint64_t)(b & 0x7F) << s2;
          if (!(b & 0x80)) { d2 = true; break; }
          s2 += 7;
        }
        if (!d2 || pos + l > len) return false;
        pos += l;
        break;
      }
      default: return false;
    }
    if (++depth_guard > 1'000'000) return false;
  }
  return true;
}

// parse HelloRequest{string name=1} without a proto library
static bool parse_hello_name(const uint8_t* p, size_t len, std::string* name) {
  size_t pos = 0;
  while (pos < len) {
    uint64_t tag = 0;
    int shift = 0;
    bool done = false;
    while (pos < len && shift <= 63) {
      uint8_t b = p[pos++];
      tag |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) { done = true; break; }
      shift += 7;
    }
    if (!done) return false;
    if ((tag >> 3) == 1 && (tag & 7) == 2) {
      uint64_t l = 0;
      int s2 = 0;
      bool d2 = false;
      while (pos < len && s2 <= 35) {
        uint8_t b = p[pos++];
        l |= (uint64_t)(b & 0x7F) << s2;
        if (!(b & 0x80)) { d2 = true; break; }
        s2 += 7;
      }
      if (!d2 || pos + l > len) return false;
      name->assign((const char*)p + pos, l);
      pos += l;
    } else {
      // skip
      uint32_t wt = tag & 7;
      if (wt == 0) {
        while (pos < len && (p[pos++] & 0x80)) {}
      } else if (wt == 1) {
        pos += 8;
      } else if (wt == 5) {
        pos += 4;
      } else if (wt == 2) {
        uint64_t l = 0;
        int s2 = 0;
        while (pos < len && s2 <= 35) {
          uint8_t b = p[pos++];
          l |= (uint64_t)(b & 0x7F) << s2;
          if (!(b & 0x80)) break;
          s2 += 7;
        }
        pos += l;
      } else {
        return false;
      }
    }
  }
  return true;
}

static std::string make_hello_response(const std::string& name) {
  std::string msg = "Hello, " + name + "!";
  std::string out;
  out.push_back('\x0a');  // field 1, wire 2
  // varint length
  uint64_t v = msg.size();
  while (v >= 0x80) {
    out.push_back((char)(v | 0x80));
    v >>= 7;
  }
  out.push_back((char)v);
  out += msg;
  return out;
}

// ---------------------------------------------------------------------------
// client
// ---------------------------------------------------------------------------

struct Call {
  std::string path;
  std::string payload;          // grpc-framed request
  size_t sent = 0;
  std::string response;         // accumulated DATA bytes
  int grpc_status = -1;         // from trailers if decodable
  std::string grpc_message;
  std::atomic<bool> done{false};
  Clock::time_point deadline;
  int slot = -1;                // caller's index
  bool no_cap = false;          // server-streaming: whole-stream bytes are
                                // unbounded by design (cap is per message)
  bool overflow = false;        // unary response exceeded max_resp
  std::vector<std::pair<std::string, std::string>> metadata;
};

// One submitted batch.  Shared ownership between the caller and the
// connection threads so an early (deadline) return never leaves dangling
// pointers; completion is signalled via the batch's cv.
struct Batch {
  std::deque<Call> calls;   // deque: stable addresses
  std::mutex mu;
  std::condition_variable cv;
  std::atomic<int> done{0};
  void complete_one() {
    { std::lock_guard<std::mutex> lk(mu); done.fetch_add(1); }
    cv.notify_one();
  }
};

struct QueuedCall {
  std::shared_ptr<Batch> batch;
  Call* call;
};

struct ClientConn {
  int fd = -1;
  int wake_fd = -1;
  nghttp2_session* sess = nullptr;
  std::thread thread;
  std::mutex mu;
  std::deque<QueuedCall> pending;              // submitted by callers
  std::unordered_map<int32_t, QueuedCall> live;
  std::string wbuf;                    // partial write buffer
  std::atomic<bool> stop{false};
  std::atomic<int> inflight{0};
  int max_inflight = 512;  // streams submitted concurrently per connection
  size_t max_resp = 0;     // unary response byte cap (0 = unlimited);
                           // reference connection.go:55-57's 4 MB recv cap
  std::string authority;
  bool broken = false;
};

static ssize_t client_data_read(nghttp2_session*, int32_t stream_id, uint8_t* buf,
                                size_t length, uint32_t* data_flags,
                                nghttp2_data_source* source, void*) {
  Call* call = (Call*)source->ptr;
  size_t left = call->payload.size() - call->sent;
  size_t n = left < length ? left : length;
  memcpy(buf, call->payload.data() + call->sent, n);
  call->sent += n;
  if (call->sent == call->payload.size()) *data_flags |= NGHTTP2_DATA_FLAG_EOF;
  return (ssize_t)n;
}

static int client_on_header(nghttp2_session* sess, const nghttp2_frame* frame,
                            const uint8_t* name, size_t namelen, const uint8_t* value,
                            size_t valuelen, uint8_t, void* user) {
  ClientConn* conn = (ClientConn*)user;
  auto it = conn->live.find(frame->hd.stream_id);
  if (it == conn->live.end()) return 0;
  if (namelen == 11 && memcmp(name, "grpc-status", 11) == 0) {
    it->second.call->grpc_status =
        atoi(std::string((const char*)value, valuelen).c_str());
  } else if (namelen == 12 && memcmp(name, "grpc-message", 12) == 0) {
    it->second.call->grpc_message.assign((const char*)value, valuelen);
  }
  return 0;
}

static int client_on_data(nghttp2_session*, uint8_t, int32_t stream_id,
                          const uint8_t* data, size_t len, void* user) {
  ClientConn* conn = (ClientConn*)user;
  auto it = conn->live.find(stream_id);
  if (it != conn->live.end()) {
    Call* c = it->second.call;
    if (!c->no_cap && conn->max_resp &&
        c->response.size() + len > conn->max_resp) {
      // cap ONE oversized unary response without failing the whole batch
      c->overflow = true;
      c->response.clear();
    } else if (!c->overflow) {
      c->response.append((const char*)data, len);
    }
  }
  return 0;
}

static int client_on_stream_close(nghttp2_session*, int32_t stream_id, uint32_t,
                                  void* user) {
  ClientConn* conn = (ClientConn*)user;
  auto it = conn->live.find(stream_id);
  if (it != conn->live.end()) {
    QueuedCall qc = it->second;
    conn->live.erase(it);
    qc.call->done.store(true, std::memory_order_release);
    conn->inflight.fetch_sub(1);
    qc.batch->complete_one();
  }
  return 0;
}

static void fail_call(ClientConn* conn, QueuedCall& qc, int status,
                      const char* msg) {
  qc.call->grpc_status = status;
  qc.call->grpc_message = msg;
  qc.call->done.store(true, std::memory_order_release);
  qc.batch->complete_one();
}


// Pump nghttp2 output to the socket until the session has nothing more to
// send or the socket blocks.  Returns false on fatal socket error; *blocked
// is set when the socket is full (caller arms POLLOUT).  The subtle case
// this exists for: a partial send() followed by a successful wbuf drain must
// LOOP BACK into nghttp2_session_mem_send — stopping there leaves queued
// frames unsent with nothing to wake the poll (observed as ~2 poll-timeout
// stalls per 1024-batch over UDS: 205 ms/batch instead of 7).
static bool flush_session(nghttp2_session* sess, int fd, std::string& wbuf,
                          bool* blocked) {
  *blocked = false;
  while (true) {
    if (!wbuf.empty()) {
      ssize_t w = send(fd, wbuf.data(), wbuf.size(), MSG_NOSIGNAL);
      if (w < 0) {
        if (errno == EAGAIN || errno == EWOULDBLOCK) { *blocked = true; return true; }
        return false;
      }
      wbuf.erase(0, (size_t)w);
      if (!wbuf.empty()) continue;
    }
    const uint8_t* out = nullptr;
    ssize_t n = nghttp2_session_mem_send(sess, &out);
    if (n < 0) return false;
    if (n == 0) return true;
    ssize_t w = send(fd, out, (size_t)n, MSG_NOSIGNAL);
    if (w < 0) {
      if (errno == EAGAIN || errno == EWOULDBLOCK) w = 0;
      else return false;
    }
    if (w < n) wbuf.assign((const char*)out + w, (size_t)(n - w));
  }
}

static void conn_loop(ClientConn* conn) {
  while (!conn->stop.load()) {
    // submit pending
    {
      std::lock_guard<std::mutex> lk(conn->mu);
      while (!conn->pending.empty() &&
             conn->inflight.load() < conn->max_inflight) {
        QueuedCall qc = conn->pending.front();
        Call* call = qc.call;
        conn->pending.pop_front();
        std::vector<nghttp2_nv> nva;
        nva.reserve(8 + call->metadata.size());
        nva.push_back(nv(":method", "POST"));
        nva.push_back(nv(":scheme", "http"));
        nva.push_back(nv(":path", call->path));
        nva.push_back(nv(":authority", conn->authority));
        nva.push_back(nv("te", "trailers"));
        nva.push_back(nv("content-type", "application/grpc"));
        for (auto& kv : call->metadata) nva.push_back(nv(kv.first, kv.second));
        nghttp2_data_provider prd;
        prd.source.ptr = call;
        prd.read_callback = client_data_read;
        int32_t sid = nghttp2_submit_request(conn->sess, nullptr, nva.data(),
                                             nva.size(), &prd, call);
        if (sid < 0) {
          fail_call(conn, qc, 14, "submit failed");  // UNAVAILABLE
          continue;
        }
        conn->live[sid] = qc;
        conn->inflight.fetch_add(1);
      }
    }
    // deadline sweep: reset overdue streams (close callback completes them)
    {
      auto now = Clock::now();
      for (auto& kv : conn->live) {
        if (kv.second.call->deadline < now && kv.second.call->grpc_status < 0) {
          kv.second.call->grpc_status = 4;  // DEADLINE_EXCEEDED
          kv.second.call->grpc_message = "deadline exceeded";
          nghttp2_submit_rst_stream(conn->sess, NGHTTP2_FLAG_NONE, kv.first,
                                    NGHTTP2_CANCEL);
        }
      }
    }
    // write
    bool write_blocked = false;
    if (!flush_session(conn->sess, conn->fd, conn->wbuf, &write_blocked))
      conn->broken = true;
    if (conn->broken) break;

    pollfd fds[2];
    fds[0] = {conn->fd, (short)(POLLIN | (write_blocked ? POLLOUT : 0)), 0};
    fds[1] = {conn->wake_fd, POLLIN, 0};
    int rc = poll(fds, 2, 100);
    if (rc < 0) break;
    if (fds[1].revents & POLLIN) {
      uint64_t junk;
      while (read(conn->wake_fd, &junk, 8) > 0) {}
    }
    if (fds[0].revents & (POLLIN | POLLERR | POLLHUP)) {
      uint8_t buf[1 << 16];
      while (true) {
        ssize_t r = recv(conn->fd, buf, sizeof(buf), 0);
        if (r > 0) {
          ssize_t consumed = nghttp2_session_mem_recv(conn->sess, buf, r);
          if (consumed < 0) { conn->broken = true; break; }
          if (r < (ssize_t)sizeof(buf)) break;
        } else if (r == 0) {
          conn->broken = true;
          break;
        } else {
          if (errno == EAGAIN || errno == EWOULDBLOCK) break;
          conn->broken = true;
          break;
        }
      }
    }
    if (conn->broken) break;
  }
  // fail any remaining calls
  std::lock_guard<std::mutex> lk(conn->mu);
  for (auto& kv : conn->live) fail_call(conn, kv.second, 14, "connection lost");
  for (auto& qc : conn->pending) fail_call(conn, qc, 14, "connection lost");
  conn->live.clear();
  conn->pending.clear();
}

class __attribute__((visibility("default"))) H2GrpcClient {
 public:
  H2GrpcClient(const std::string& target, int n_connections,
               const std::string& authority, int max_inflight = 512,
               size_t max_resp_bytes = 0)
      : target_(target) {
    if (n_connections < 1) n_connections = 1;
    for (int i = 0; i < n_connections; ++i) {
      auto conn = std::make_unique<ClientConn>();
      conn->max_inflight = max_inflight < 1 ? 1 : max_inflight;
      conn->max_resp = max_resp_bytes;
      conn->fd = connect_target(target);
      conn->wake_fd = eventfd(0, EFD_NONBLOCK);
      conn->authority = authority.empty() ? "localhost" : authority;

      nghttp2_session_callbacks* cbs;
      nghttp2_session_callbacks_new(&cbs);
      nghttp2_session_callbacks_set_on_header_callback(cbs, client_on_header);
      nghttp2_session_callbacks_set_on_data_chunk_recv_callback(cbs, client_on_data);
      nghttp2_session_callbacks_set_on_stream_close_callback(cbs, client_on_stream_close);
      nghttp2_session_client_new(&conn->sess, cbs, conn.get());
      nghttp2_session_callbacks_del(cbs);

      nghttp2_settings_entry iv[] = {
          {NGHTTP2_SETTINGS_INITIAL_WINDOW_SIZE, (1u << 30)},
          {NGHTTP2_SETTINGS_MAX_CONCURRENT_STREAMS, 8192},
          {NGHTTP2_SETTINGS_MAX_FRAME_SIZE, 1u << 20},
      };
      nghttp2_submit_settings(conn->sess, NGHTTP2_FLAG_NONE, iv, 3);
      nghttp2_session_set_local_window_size(conn->sess, NGHTTP2_FLAG_NONE, 0, 1 << 30);

      ClientConn* raw = conn.get();
      conn->thread = std::thread([raw] { conn_loop(raw); });
      conns_.push_back(std::move(conn));
    }
  }

  ~H2GrpcClient() { close_all(); }

  void close_all() {
    for (auto& conn : conns_) {
      conn->stop.store(true);
      uint64_t one = 1;
      (void)!write(conn->wake_fd, &one, 8);
      if (conn->thread.joinable()) conn->thread.join();
      if (conn->sess) nghttp2_session_del(conn->sess);
      if (conn->fd >= 0) close(conn->fd);
      if (conn->wake_fd >= 0) close(conn->wake_fd);
      conn->sess = nullptr;
      conn->fd = conn->wake_fd = -1;
    }
    conns_.clear();
  }

  bool healthy() const {
    for (auto& conn : conns_)
      if (!conn->broken) return true;
    return false;
  }

  // Batch server-streaming: like invoke_batch but each call returns ALL its
  // messages as ONE contiguous buffer + per-message lengths:
  // (grpc_status, payload_blob, [len, ...], message).  One bytes object per
  // STREAM instead of one per message — at 4096 msgs/stream the per-object
  // allocation cost dominated the step otherwise.
  std::vector<std::tuple<int, py::bytes, std::vector<uint32_t>, std::string>>
  invoke_stream_batch(
      const std::vector<std::string>& paths, const std::vector<py::bytes>& payloads,
      double timeout_s,
      const std::vector<std::vector<std::pair<std::string, std::string>>>& metadata) {
    auto raw = invoke_collect(paths, payloads, timeout_s, metadata, /*no_cap=*/true);
    std::vector<std::tuple<int, py::bytes, std::vector<uint32_t>, std::string>> out;
    out.reserve(raw.size());
    std::string blob;
    for (auto& r : raw) {
      int status = std::get<0>(r);
      const std::string& data = std::get<1>(r);
      std::vector<uint32_t> lens;
      blob.clear();
      blob.reserve(data.size());
      size_t pos = 0;
      bool compressed = false;
      while (pos + 5 <= data.size()) {
        if (data[pos] != 0) {  // compressed-flag frame: not negotiated
          compressed = true;
          break;
        }
        uint32_t len;
        memcpy(&len, data.data() + pos + 1, 4);
        len = ntohl(len);
        if (pos + 5 + len > data.size()) break;
        blob.append(data, pos + 5, len);
        lens.push_back(len);
        pos += 5 + len;
      }
      if (status == 0 && compressed) {
        status = 13;  // INTERNAL
        std::get<2>(r) = "compressed gRPC response frame not supported";
      }
      // No decodable grpc-status trailer = malformed close: surface it even
      // when DATA frames arrived (never silently treat a half-delivered
      // stream as complete).
      if (status < 0) {
        status = 2;  // UNKNOWN
        if (std::get<2>(r).empty())
          std::get<2>(r) = "stream closed without grpc-status";
      }
      out.emplace_back(status, py::bytes(blob), std::move(lens), std::get<2>(r));
    }
    return out;
  }

  // Batch unary: returns list of (grpc_status, payload, message).
  // status -1 from the wire means "closed without decodable grpc-status":
  // always surfaced as UNKNOWN (gRPC requires trailers).
  std::vector<std::tuple<int, py::bytes, std::string>> invoke_batch(
      const std::vector<std::string>& paths, const std::vector<py::bytes>& payloads,
      double timeout_s,
      const std::vector<std::vector<std::pair<std::string, std::string>>>& metadata) {
    auto raw = invoke_collect(paths, payloads, timeout_s, metadata);
    std::vector<std::tuple<int, py::bytes, std::string>> out;
    out.reserve(raw.size());
    for (auto& r : raw) {
      int status = std::get<0>(r);
      const std::string& data = std::get<1>(r);
      std::string payload;
      bool compressed = false;
      if (data.size() >= 5) {
        compressed = data[0] != 0;
        uint32_t len;
        memcpy(&len, data.data() + 1, 4);
        len = ntohl(len);
        if (data.size() >= 5 + (size_t)len) payload = data.substr(5, len);
      }
      if (status < 0) {
        // malformed close (no grpc-status trailer): an error, even if a
        // full message arrived — gRPC requires trailers
        status = 2;  // UNKNOWN
        if (std::get<2>(r).empty())
          std::get<2>(r) = "stream closed without grpc-status";
      } else if (status == 0 && compressed) {
        status = 13;  // INTERNAL
        std::get<2>(r) = "compressed gRPC response frame not supported";
        payload.clear();
      }
      out.emplace_back(status, py::bytes(payload), std::get<2>(r));
    }
    return out;
  }

 public:
  // Raw-pointer batch submit for in-process callers (the engine's native
  // span executor): payloads point into pinned staging memory; call with
  // the GIL RELEASED.  Returns (grpc_status, DATA bytes, message) per call.
  struct RawCall {
    const char* path;
    size_t path_len;
    const uint8_t* payload;
    size_t payload_len;
    const std::vector<std::pair<std::string, std::string>>* metadata;  // may be null
  };
  std::vector<std::tuple<int, std::string, std::string>> invoke_raw(
      const std::vector<RawCall>& rc, double timeout_s) {
    size_t n = rc.size();
    auto batch = std::make_shared<Batch>();
    auto deadline = Clock::now() + std::chrono::duration_cast<Clock::duration>(
                                       std::chrono::duration<double>(timeout_s));
    for (size_t i = 0; i < n; ++i) {
      batch->calls.emplace_back();
      Call& call = batch->calls.back();
      call.path.assign(rc[i].path, rc[i].path_len);
      call.deadline = deadline;
      call.slot = (int)i;
      if (rc[i].metadata) call.metadata = *rc[i].metadata;
      call.payload = grpc_frame(
          std::string((const char*)rc[i].payload, rc[i].payload_len));
    }
    size_t nc = conns_.size();
    for (size_t i = 0; i < n; ++i) {
      ClientConn* conn = conns_[i % nc].get();
      std::lock_guard<std::mutex> lk(conn->mu);
      conn->pending.push_back(QueuedCall{batch, &batch->calls[i]});
    }
    for (auto& conn : conns_) {
      uint64_t one = 1;
      (void)!write(conn->wake_fd, &one, 8);
    }
    {
      std::unique_lock<std::mutex> lk(batch->mu);
      auto hard = deadline + std::chrono::seconds(2);
      while (batch->done.load() < (int)n) {
        if (batch->cv.wait_until(lk, hard) == std::cv_status::timeout) break;
      }
    }
    std::vector<std::tuple<int, std::string, std::string>> out;
    out.reserve(n);
    for (size_t i = 0; i < n; ++i) {
      Call& call = batch->calls[i];
      if (!call.done.load(std::memory_order_acquire)) {
        out.emplace_back(4, std::string(), "deadline exceeded");
        continue;
      }
      if (call.overflow) {  // RESOURCE_EXHAUSTED, this slot only
        out.emplace_back(8, std::string(),
                         "response exceeds max_recv_msg_bytes");
        continue;
      }
      out.emplace_back(call.grpc_status, std::move(call.response),
                       call.grpc_message);
    }
    return out;
  }

 private:
  // submit the batch, wait for completion, return raw DATA byte streams
  std::vector<std::tuple<int, std::string, std::string>> invoke_collect(
      const std::vector<std::string>& paths, const std::vector<py::bytes>& payloads,
      double timeout_s,
      const std::vector<std::vector<std::pair<std::string, std::string>>>& metadata,
      bool no_cap = false) {
    size_t n = paths.size();
    if (payloads.size() != n) throw std::runtime_error("paths/payloads mismatch");
    auto batch = std::make_shared<Batch>();
    auto deadline = Clock::now() + std::chrono::duration_cast<Clock::duration>(
                                       std::chrono::duration<double>(timeout_s));
    for (size_t i = 0; i < n; ++i) {
      batch->calls.emplace_back();
      Call& call = batch->calls.back();
      call.path = paths[i];
      call.deadline = deadline;
      call.slot = (int)i;
      call.no_cap = no_cap;
      if (i < metadata.size()) call.metadata = metadata[i];
      std::string raw = payloads[i];  // needs GIL; held here
      call.payload = grpc_frame(raw);
    }
    {
      py::gil_scoped_release rel;
      size_t nc = conns_.size();
      for (size_t i = 0; i < n; ++i) {
        ClientConn* conn = conns_[i % nc].get();
        std::lock_guard<std::mutex> lk(conn->mu);
        conn->pending.push_back(QueuedCall{batch, &batch->calls[i]});
      }
      for (auto& conn : conns_) {
        uint64_t one = 1;
        (void)!write(conn->wake_fd, &one, 8);
      }
      std::unique_lock<std::mutex> lk(batch->mu);
      // grace past the gRPC deadline so DEADLINE_EXCEEDED resolves cleanly
      auto hard = deadline + std::chrono::seconds(2);
      while (batch->done.load() < (int)n) {
        if (batch->cv.wait_until(lk, hard) == std::cv_status::timeout) break;
      }
    }
    std::vector<std::tuple<int, std::string, std::string>> out;
    out.reserve(n);
    for (size_t i = 0; i < n; ++i) {
      Call& call = batch->calls[i];
      if (!call.done.load(std::memory_order_acquire)) {
        out.emplace_back(4, std::string(), "deadline exceeded");
        continue;
      }
      if (call.overflow) {  // RESOURCE_EXHAUSTED, this slot only
        out.emplace_back(8, std::string(),
                         "response exceeds max_recv_msg_bytes");
        continue;
      }
      out.emplace_back(call.grpc_status, std::move(call.response),
                       call.grpc_message);
    }
    return out;
  }

 public:

 private:
  std::string target_;
  std::vector<std::unique_ptr<ClientConn>> conns_;
};

// ---------------------------------------------------------------------------
// server (bench backend)
// ---------------------------------------------------------------------------

struct ServerStream {
  std::string path;
  std::string body;
  std::string response;  // grpc-framed
  size_t sent = 0;
  int grpc_status = 0;
  std::string grpc_message;
  bool responded = false;
};

struct ServerConn {
  int fd = -1;
  nghttp2_session* sess = nullptr;
  std::unordered_map<int32_t, ServerStream> streams;
  std::string wbuf;
  bool broken = false;
  class H2Server* server = nullptr;
};

class H2Server {
 public:
  // kind: 0 = echo+validate, 1 = hello, 2 = stream_echo (server-streaming:
  // emits N copies of the request message, N = Wide64.f02_int32, the shape
  // of BASELINE config 4)
  std::unordered_map<std::string, int> routes;

  H2Server(const std::string& target) : target_(target) {}

  ~H2Server() { stop(); }

  void add_route(const std::string& path, const std::string& kind) {
    routes[path] = (kind == "hello") ? 1 : (kind == "stream_echo") ? 2 : 0;
  }

  std::string start() {
    if (target_.rfind("unix:", 0) == 0) {
      listen_fd_ = socket(AF_UNIX, SOCK_STREAM, 0);
      sockaddr_un addr{};
      addr.sun_family = AF_UNIX;
      std::string path = target_.substr(5);
      if (path.size() >= sizeof(addr.sun_path))
        throw std::runtime_error("uds path too long: " + path);
      unlink(path.c_str());
      memcpy(addr.sun_path, path.c_str(), path.size() + 1);
      if (bind(listen_fd_, (sockaddr*)&addr, sizeof(addr)) != 0)
        throw std::runtime_error("bind failed: " + target_);
      bound_ = target_;
    } else {
      auto colon = target_.rfind(':');
      std::string host = target_.substr(0, colon);
      int port = std::stoi(target_.substr(colon + 1));
      listen_fd_ = socket(AF_INET, SOCK_STREAM, 0);
      int one = 1;
      setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
      sockaddr_in addr{};
      addr.sin_family = AF_INET;
      addr.sin_port = htons((uint16_t)port);
      inet_pton(AF_INET, host.c_str(), &addr.sin_addr);
      if (bind(listen_fd_, (sockaddr*)&addr, sizeof(addr)) != 0)
        throw std::runtime_error("bind failed: " + target_);
      socklen_t alen = sizeof(addr);
      getsockname(listen_fd_, (sockaddr*)&addr, &alen);
      bound_ = host + ":" + std::to_string(ntohs(addr.sin_port));
    }
    if (listen(listen_fd_, 512) != 0) throw std::runtime_error("listen failed");
    int flags = fcntl(listen_fd_, F_GETFL, 0);
    fcntl(listen_fd_, F_SETFL, flags | O_NONBLOCK);
    stop_.store(false);
    accept_thread_ = std::thread([this] { accept_loop(); });
    return bound_;
  }

  void stop() {
    if (stop_.exchange(true)) return;
    if (accept_thread_.joinable()) accept_thread_.join();
    if (listen_fd_ >= 0) close(listen_fd_);
    listen_fd_ = -1;
  }

  // total requests handled (all routes) — lets tests assert exactly-once
  // invocation semantics of the gateway's fallback paths
  long request_count() const { return requests_.load(); }

  void handle(ServerStream& st) {
    requests_.fetch_add(1, std::memory_order_relaxed);
    auto it = routes.find(st.path);
    if (it == routes.end()) {
      st.grpc_status = 12;  // UNIMPLEMENTED
      st.grpc_message = "unknown method " + st.path;
      return;
    }
    if (st.body.size() < 5) {
      st.grpc_status = 13;
      st.grpc_message = "truncated grpc frame";
      return;
    }
    uint32_t len;
    memcpy(&len, st.body.data() + 1, 4);
    len = ntohl(len);
    if (st.body.size() < 5 + (size_t)len) {
      st.grpc_status = 13;
      st.grpc_message = "short grpc frame";
      return;
    }
    const uint8_t* msg = (const uint8_t*)st.body.data() + 5;
    if (it->second == 2) {
      // server-streaming echo: field 2 varint = message count
      if (!pb_validate(msg, len)) {
        st.grpc_status = 13;
        st.grpc_message = "malformed protobuf";
        return;
      }
      int64_t count = 1;
      {
        size_t pos = 0;
        while (pos < len) {
          uint64_t tag = 0;
          int shift = 0;
          while (pos < len && shift <= 63) {
            uint8_t b = msg[pos++];
            tag |= (uint64_t)(b & 0x7F) << shift;
            if (!(b & 0x80)) break;
            shift += 7;
          }
          uint32_t num = (uint32_t)(tag >> 3), wt = (uint32_t)(tag & 7);
          if (num == 2 && wt == 0) {
            uint64_t v = 0;
            int s2 = 0;
            while (pos < len && s2 <= 63) {
              uint8_t b = msg[pos++];
              v |= (uint64_t)(b & 0x7F) << s2;
              if (!(b & 0x80)) break;
              s2 += 7;
            }
            count = (int64_t)(int32_t)(uint32_t)v;
            break;
          }
          // skip
          if (wt == 0) {
            while (pos < len && (msg[pos++] & 0x80)) {}
          } else if (wt == 1) {
            pos += 8;
          } else if (wt == 5) {
            pos += 4;
          } else if (wt == 2) {
            uint64_t l = 0;
            int s2 = 0;
            while (pos < len && s2 <= 35) {
              uint8_t b = msg[pos++];
              l |= (uint64_t)(b & 0x7F) << s2;
              if (!(b & 0x80)) break;
              s2 += 7;
            }
            pos += l;
          } else {
            break;
          }
        }
      }
      if (count < 1) count = 1;
      if (count > 65536) count = 65536;
      std::string one = grpc_frame(std::string((const char*)msg, len));
      st.response.reserve(one.size() * count);
      for (int64_t i = 0; i < count; ++i) st.response += one;
      return;
    }
    if (it->second == 1) {
      std::string name;
      if (!parse_hello_name(msg, len, &name)) {
        st.grpc_status = 13;
        st.grpc_message = "bad HelloRequest";
        return;
      }
      if (name == "error") {
        st.grpc_status = 3;  // INVALID_ARGUMENT (mirrors the demo backend)
        st.grpc_message = "name must not be 'error'";
        return;
      }
      if (name == "badutf8") {
        // hostile-backend fixture: a HelloResponse whose message field
        // contains invalid UTF-8 (0xFF 0xFE).  The GPU decode stage must
        // reject it and the gateway must transcode the received bytes on
        // the host WITHOUT re-invoking (request_count() stays at 1).
        std::string msg = "bad\xff\xfe";
        std::string out;
        out.push_back('\x0a');
        out.push_back((char)msg.size());
        out += msg;
        st.response = grpc_frame(out);
        return;
      }
      st.response = grpc_frame(make_hello_response(name));
    } else {
      if (!pb_validate(msg, len)) {
        st.grpc_status = 13;
        st.grpc_message = "malformed protobuf";
        return;
      }
      st.response = grpc_frame(std::string((const char*)msg, len));
    }
  }

 private:
  void accept_loop() {
    std::vector<std::thread> workers;
    while (!stop_.load()) {
      pollfd pfd{listen_fd_, POLLIN, 0};
      int rc = poll(&pfd, 1, 200);
      if (rc <= 0) continue;
      int cfd = accept(listen_fd_, nullptr, nullptr);
      if (cfd < 0) continue;
      int flags = fcntl(cfd, F_GETFL, 0);
      fcntl(cfd, F_SETFL, flags | O_NONBLOCK);
      int one = 1;
      setsockopt(cfd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      apply_sockbuf(cfd);
      workers.emplace_back([this, cfd] { conn_loop(cfd); });
    }
    for (auto& t : workers)
      if (t.joinable()) t.join();
  }

  static ssize_t resp_read(nghttp2_session* sess, int32_t stream_id, uint8_t* buf,
                           size_t length, uint32_t* data_flags,
                           nghttp2_data_source* source, void*) {
    ServerStream* st = (ServerStream*)source->ptr;
    size_t left = st->response.size() - st->sent;
    size_t n = left < length ? left : length;
    memcpy(buf, st->response.data() + st->sent, n);
    st->sent += n;
    if (st->sent == st->response.size()) {
      *data_flags |= NGHTTP2_DATA_FLAG_EOF | NGHTTP2_DATA_FLAG_NO_END_STREAM;
      std::string status = std::to_string(st->grpc_status);
      std::vector<nghttp2_nv> trailers;
      trailers.push_back(nv("grpc-status", status));
      if (!st->grpc_message.empty())
        trailers.push_back(nv("grpc-message", st->grpc_message));
      nghttp2_submit_trailer(sess, stream_id, trailers.data(), trailers.size());
    }
    return (ssize_t)n;
  }

  void respond(ServerConn* conn, int32_t sid, ServerStream& st) {
    handle(st);
    std::string status = std::to_string(st.grpc_status);
    if (st.grpc_status != 0 || st.response.empty()) {
      // trailers-only response
      std::vector<nghttp2_nv> hdrs;
      hdrs.push_back(nv(":status", "200"));
      hdrs.push_back(nv("content-type", "application/grpc"));
      hdrs.push_back(nv("grpc-status", status));
      if (!st.grpc_message.empty())
        hdrs.push_back(nv("grpc-message", st.grpc_message));
      nghttp2_submit_response(conn->sess, sid, hdrs.data(), hdrs.size(), nullptr);
      return;
    }
    nghttp2_nv hdrs[] = {nv(":status", "200"), nv("content-type", "application/grpc")};
    nghttp2_data_provider prd;
    prd.source.ptr = &st;
    prd.read_callback = resp_read;
    nghttp2_submit_response(conn->sess, sid, hdrs, 2, &prd);
  }

  static int s_on_begin_headers(nghttp2_session*, const nghttp2_frame* frame, void* user) {
    ServerConn* conn = (ServerConn*)user;
    if (frame->hd.type == NGHTTP2_HEADERS &&
        frame->headers.cat == NGHTTP2_HCAT_REQUEST)
      conn->streams[frame->hd.stream_id];
    return 0;
  }

  static int s_on_header(nghttp2_session*, const nghttp2_frame* frame,
                         const uint8_t* name, size_t namelen, const uint8_t* value,
                         size_t valuelen, uint8_t, void* user) {
    ServerConn* conn = (ServerConn*)user;
    auto it = conn->streams.find(frame->hd.stream_id);
    if (it == conn->streams.end()) return 0;
    if (namelen == 5 && memcmp(name, ":path", 5) == 0)
      it->second.path.assign((const char*)value, valuelen);
    return 0;
  }

  static int s_on_data(nghttp2_session*, uint8_t, int32_t stream_id,
                       const uint8_t* data, size_t len, void* user) {
    ServerConn* conn = (ServerConn*)user;
    auto it = conn->streams.find(stream_id);
    if (it != conn->streams.end()) it->second.body.append((const char*)data, len);
    return 0;
  }

  static int s_on_frame_recv(nghttp2_session*, const nghttp2_frame* frame, void* user) {
    ServerConn* conn = (ServerConn*)user;
    if ((frame->hd.type == NGHTTP2_DATA || frame->hd.type == NGHTTP2_HEADERS) &&
        (frame->hd.flags & NGHTTP2_FLAG_END_STREAM)) {
      auto it = conn->streams.find(frame->hd.stream_id);
      if (it != conn->streams.end() && !it->second.responded) {
        it->second.responded = true;
        conn->server->respond(conn, frame->hd.stream_id, it->second);
      }
    }
    return 0;
  }

  static int s_on_stream_close(nghttp2_session*, int32_t stream_id, uint32_t,
                               void* user) {
    ServerConn* conn = (ServerConn*)user;
    conn->streams.erase(stream_id);
    return 0;
  }

  void conn_loop(int cfd) {
    ServerConn conn;
    conn.fd = cfd;
    conn.server = this;
    nghttp2_session_callbacks* cbs;
    nghttp2_session_callbacks_new(&cbs);
    nghttp2_session_callbacks_set_on_begin_headers_callback(cbs, s_on_begin_headers);
    nghttp2_session_callbacks_set_on_header_callback(cbs, s_on_header);
    nghttp2_session_callbacks_set_on_data_chunk_recv_callback(cbs, s_on_data);
    nghttp2_session_callbacks_set_on_frame_recv_callback(cbs, s_on_frame_recv);
    nghttp2_session_callbacks_set_on_stream_close_callback(cbs, s_on_stream_close);
    nghttp2_session_server_new(&conn.sess, cbs, &conn);
    nghttp2_session_callbacks_del(cbs);
    nghttp2_settings_entry iv[] = {
        {NGHTTP2_SETTINGS_INITIAL_WINDOW_SIZE, (1u << 30)},
        {NGHTTP2_SETTINGS_MAX_CONCURRENT_STREAMS, 8192},
        {NGHTTP2_SETTINGS_MAX_FRAME_SIZE, 1u << 20},
    };
    nghttp2_submit_settings(conn.sess, NGHTTP2_FLAG_NONE, iv, 3);
    nghttp2_session_set_local_window_size(conn.sess, NGHTTP2_FLAG_NONE, 0, 1 << 30);

    while (!stop_.load() && !conn.broken) {
      bool write_blocked = false;
      if (!flush_session(conn.sess, conn.fd, conn.wbuf, &write_blocked))
        conn.broken = true;
      if (conn.broken) break;
      pollfd pfd{conn.fd, (short)(POLLIN | (write_blocked ? POLLOUT : 0)), 0};
      int rc = poll(&pfd, 1, 200);
      if (rc < 0) break;
      if (pfd.revents & (POLLIN | POLLERR | POLLHUP)) {
        uint8_t buf[1 << 16];
        while (true) {
          ssize_t r = recv(conn.fd, buf, sizeof(buf), 0);
          if (r > 0) {
            if (nghttp2_session_mem_recv(conn.sess, buf, r) < 0) {
              conn.broken = true;
              break;
            }
            if (r < (ssize_t)sizeof(buf)) break;
          } else if (r == 0) {
            conn.broken = true;
            break;
          } else {
            if (errno == EAGAIN || errno == EWOULDBLOCK) break;
            conn.broken = true;
            break;
          }
        }
      }
      if (nghttp2_session_want_read(conn.sess) == 0 &&
          nghttp2_session_want_write(conn.sess) == 0)
        break;
    }
    nghttp2_session_del(conn.sess);
    close(cfd);
  }

  std::string target_;
  std::string bound_;
  int listen_fd_ = -1;
  std::thread accept_thread_;
  std::atomic<bool> stop_{true};
  std::atomic<long> requests_{0};
};

