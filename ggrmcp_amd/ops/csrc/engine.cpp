// _jsonproto — host runtime for the MI355X batch-transcode engine.
//
// Owns the device arenas (HBM), pinned host staging buffers and the HIP
// stream for one engine instance; uploads the flat descriptor tables
// compiled by ggrmcp_amd/engine/tables.py; launches the two kernels
// (k_json2pb ingest, k_pb2json respond) over whole batches.  The reference
// gateway has no equivalent — its hot path is per-request Go reflection
// (aalobaidi/ggRMCP pkg/grpc/reflection.go:333-391).
//
// Single compilation unit: the kernels are #included below so no
// relocatable-device-code link step is needed.  Build: ops/build.py
// (hipcc --offload-arch=gfx950).
//
// Concurrency model: one Engine == one HIP stream == one in-flight batch.
// encode*/decode* release the GIL while the GPU works, so Python-side
// pipeline threads (gRPC I/O) overlap with kernels; multiple Engine
// instances on the same device give copy/compute overlap across batches.
//
// Two call styles per direction:
//   encode(data, in_off, pb_off, ...)  — caller-staged contiguous buffer
//   encode_list([bytes, ...], ...)     — C++ gathers the Python list into
//     pinned memory and builds the offsets (removes the join/slice loops
//     from the Python hot path); same for decode/decode_list.

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>

#include <chrono>
#include <cstring>
#include <stdexcept>
#include <string>

#include "h2grpc_impl.h"
#include "json2pb.hip"
#include "pb2json.hip"
#include "span_api.h"

namespace py = pybind11;

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess)                                                   \
      throw std::runtime_error(std::string("HIP error at " #expr ": ") +    \
                               hipGetErrorString(_e));                      \
  } while (0)

namespace {

constexpr int PIPE_MAX_CHUNKS = 8;

struct DeviceBuf {
  void* p = nullptr;
  size_t n = 0;
  void alloc(size_t bytes) {
    HIP_CHECK(hipMalloc(&p, bytes));
    n = bytes;
  }
  ~DeviceBuf() {
    if (p) (void)hipFree(p);
  }
};

struct PinnedBuf {
  void* p = nullptr;
  size_t n = 0;
  void alloc(size_t bytes) {
    HIP_CHECK(hipHostMalloc(&p, bytes, hipHostMallocDefault));
    n = bytes;
  }
  ~PinnedBuf() {
    if (p) (void)hipHostFree(p);
  }
};

uint64_t cdiv(uint64_t a, uint64_t b) { return (a + b - 1) / b; }
inline size_t align64(size_t x) { return (x + 63) & ~(size_t)63; }

// non-owning window into a packed DeviceBuf/PinnedBuf allocation
struct BufView {
  void* p = nullptr;
  size_t n = 0;
};

// workgroup-decode routing threshold; GGRMCP_WG_DEC_MIN overrides for
// classic-vs-cooperative differential tests (read per batch: ~100 ns)
inline uint32_t wg_dec_min() {
  const char* e = getenv("GGRMCP_WG_DEC_MIN");
  return e ? (uint32_t)strtoul(e, nullptr, 10) : WG_DEC_MIN_BYTES;
}

inline uint32_t wg_enc_min() {
  const char* e = getenv("GGRMCP_WG_ENC_MIN");
  return e ? (uint32_t)strtoul(e, nullptr, 10) : WG_ENC_MIN_BYTES;
}

// debug bisection: limit the wg kernels to their first N phases
inline int wg_phases() {
  const char* e = getenv("GGRMCP_WG_PHASES");
  return e ? atoi(e) : 3;
}
inline int wg_enc_phases() {
  const char* e = getenv("GGRMCP_WG_ENC_PHASES");
  return e ? atoi(e) : 3;
}
// multi-wave speculative structural scan in the wg encode phase A
// (GGRMCP_MW_SCAN=0 disables).  Default ON: byte-identical across the
// differential suites and +9% wide64 end-to-end (encode span -25%,
// profiles/mw_w64_d2.json); spans < 8 KB keep the serial walk.
inline int mw_scan_on() {
  const char* e = getenv("GGRMCP_MW_SCAN");
  return e ? atoi(e) : 1;
}

// borrow (ptr, len) from a bytes / bytearray / memoryview element
inline bool view_of(py::handle el, const char** ptr, size_t* len) {
  if (PyBytes_Check(el.ptr())) {
    *ptr = PyBytes_AS_STRING(el.ptr());
    *len = (size_t)PyBytes_GET_SIZE(el.ptr());
    return true;
  }
  if (el.is_none()) {
    *ptr = nullptr;
    *len = 0;
    return true;
  }
  Py_buffer view;
  if (PyObject_GetBuffer(el.ptr(), &view, PyBUF_CONTIG_RO) != 0) {
    PyErr_Clear();
    return false;
  }
  *ptr = (const char*)view.buf;
  *len = (size_t)view.len;
  PyBuffer_Release(&view);  // borrowed pointers stay valid for bytes-likes
  return true;
}

// per-request arena sizing rules (single source of truth, mirrored nowhere)
// +25% headroom: 3-byte non-minimal length slots exceed JSON syntax
// overhead for small-entry maps/arrays (entry+key+value slots ~9B per
// entry vs ~4 chars of JSON punctuation)
inline size_t pb_cap(size_t in_len) {
  return (in_len + in_len / 4 + 192 + 15) & ~(size_t)15;
}
inline size_t scratch_cap(size_t wire_len) {
  // +WG_DEC_EXTRA for large responses: the workgroup-cooperative decode
  // pads each top-level item's scratch region (common.h); keyed on the
  // same (env-overridable) threshold the routing uses
  return (wire_len * 8 + 1024 +
          (wire_len >= wg_dec_min() ? (size_t)WG_DEC_EXTRA : 0) + 15) &
         ~(size_t)15;
}
inline size_t final_cap(size_t wire_len) {
  return (wire_len * 16 + 2048 + 15) & ~(size_t)15;
}

// JSON string escaping for C++-assembled error envelopes (error details may
// contain quotes/control bytes)
void json_escape_append(std::string& out, const char* s, size_t n) {
  static const char* hex = "0123456789abcdef";
  for (size_t i = 0; i < n; ++i) {
    unsigned char c = (unsigned char)s[i];
    if (c == '"' || c == '\\') {
      out.push_back('\\');
      out.push_back((char)c);
    } else if (c == '\n') {
      out += "\\n";
    } else if (c == '\r') {
      out += "\\r";
    } else if (c == '\t') {
      out += "\\t";
    } else if (c < 0x20) {
      out += "\\u00";
      out.push_back(hex[c >> 4]);
      out.push_back(hex[c & 15]);
    } else {
      out.push_back((char)c);
    }
  }
}

// grpc status code -> canonical name (mirrors backend/native_invoker.py
// _CODE_NAMES so native and Python error texts match)
const char* grpc_status_name(int s) {
  static const char* names[] = {
      "OK", "CANCELLED", "UNKNOWN", "INVALID_ARGUMENT", "DEADLINE_EXCEEDED",
      "NOT_FOUND", "ALREADY_EXISTS", "PERMISSION_DENIED", "RESOURCE_EXHAUSTED",
      "FAILED_PRECONDITION", "ABORTED", "OUT_OF_RANGE", "UNIMPLEMENTED",
      "INTERNAL", "UNAVAILABLE", "DATA_LOSS", "UNAUTHENTICATED"};
  return (s >= 0 && s < 17) ? names[s] : "CODE_?";
}

// kernel status -> JSON-RPC (code, message); mirrors engine/batch.py
// _STATUS_TO_RPC (reference error mapping handler.go:117-127)
void status_to_rpc(int status, int* code, const char** msg) {
  switch (status) {
    case E_PARSE: *code = -32700; *msg = "parse error"; break;
    case E_INVALID_REQUEST: *code = -32600; *msg = "invalid request"; break;
    case E_METHOD_NOT_FOUND: *code = -32601; *msg = "tool not found"; break;
    case E_INVALID_PARAMS: *code = -32602; *msg = "invalid params"; break;
    case E_LIMIT: *code = -32602; *msg = "argument limits exceeded"; break;
    default: *code = -32603; *msg = "internal error"; break;
  }
}

}  // namespace

// Parallel staging: the span executor's per-batch memcpys into pinned
// arenas are single-threaded (~10-15 GB/s -> ~0.4 ms per 4 MB wide-
// payload batch, each direction).  A tiny per-engine helper pool splits
// the slot range by bytes for big batches; small batches stay inline
// (condvar wakeups cost more than the copy).  GGRMCP_STAGE_THREADS
// helpers (default 3, 0 disables), engages at GGRMCP_STAGE_MIN bytes
// (default 1 MiB).
class StagePool {
 public:
  explicit StagePool(int helpers) {
    for (int i = 0; i < helpers; ++i) th_.emplace_back([this] { loop(); });
  }
  ~StagePool() {
    {
      std::lock_guard<std::mutex> lk(mu_);
      stop_ = true;
    }
    cv_.notify_all();
    for (auto& t : th_) t.join();
  }
  // run fn(i) for i in [0, k): caller participates, helpers join in
  void run(int k, const std::function<void(int)>& fn) {
    if (k <= 1 || th_.empty()) {
      for (int i = 0; i < k; ++i) fn(i);
      return;
    }
    {
      std::lock_guard<std::mutex> lk(mu_);
      fn_ = &fn;
      next_ = 0;
      total_ = k;
      done_ = 0;
      ++gen_;
    }
    cv_.notify_all();
    work();
    std::unique_lock<std::mutex> lk(mu_);
    fin_cv_.wait(lk, [this] { return done_ == total_; });
    fn_ = nullptr;
  }

 private:
  void work() {
    while (true) {
      int i;
      {
        std::lock_guard<std::mutex> lk(mu_);
        if (!fn_ || next_ >= total_) return;
        i = next_++;
      }
      (*fn_)(i);
      {
        std::lock_guard<std::mutex> lk(mu_);
        if (++done_ == total_) fin_cv_.notify_all();
      }
    }
  }
  void loop() {
    uint64_t seen = 0;
    while (true) {
      {
        std::unique_lock<std::mutex> lk(mu_);
        cv_.wait(lk, [&] {
          return stop_ || (fn_ && gen_ != seen && next_ < total_);
        });
        if (stop_) return;
        seen = gen_;
      }
      work();
    }
  }
  std::vector<std::thread> th_;
  std::mutex mu_;
  std::condition_variable cv_, fin_cv_;
  const std::function<void(int)>* fn_ = nullptr;
  int next_ = 0, total_ = 0, done_ = 0;
  uint64_t gen_ = 0;
  bool stop_ = false;
};

inline int stage_threads() {
  // default 0 (off): measured NEUTRAL on wide64 (51.9k with 3 helpers vs
  // 53.0k without, same box) — the engines already stage in parallel
  // across instances and the pool wakeups eat the ~0.2-0.4 ms the split
  // saves.  Kept env-gated for single-engine deployments.
  const char* e = getenv("GGRMCP_STAGE_THREADS");
  int v = e ? atoi(e) : 0;
  return v < 0 ? 0 : (v > 15 ? 15 : v);
}
inline size_t stage_min_bytes() {
  const char* e = getenv("GGRMCP_STAGE_MIN");
  return e ? (size_t)atoll(e) : (size_t)(1u << 20);
}

class Engine : public spanapi::ISpanExecutor {
 public:
  Engine(int device, py::bytes msg_table, py::bytes field_table,
         py::bytes enum_table, py::bytes enum_values, py::bytes tool_table,
         py::bytes name_blob, int n_msgs, int n_tools, int max_batch,
         size_t cap_in, size_t cap_pb, size_t cap_scratch, size_t cap_final,
         py::object tool_paths, py::object tool_out_msg,
         py::object tool_backend)
      : device_(device), max_batch_(max_batch) {
    if (!tool_paths.is_none()) {
      for (auto el : tool_paths.cast<py::sequence>())
        tool_paths_.push_back(el.cast<std::string>());
    }
    if (!tool_out_msg.is_none()) {
      auto arr = tool_out_msg.cast<py::array_t<int32_t>>();
      tool_out_msg_.assign(arr.data(), arr.data() + arr.size());
    }
    if (!tool_backend.is_none()) {
      auto arr = tool_backend.cast<py::array_t<int32_t>>();
      tool_backend_.assign(arr.data(), arr.data() + arr.size());
    }
    HIP_CHECK(hipSetDevice(device_));
    // Both kernels are ITERATIVE (explicit MAX_RECURSE-deep frame stacks in
    // decode_walk/encode_walk): the private segment is statically sized by
    // the compiler, so no hipLimitStackSize raise is needed — the earlier
    // recursive versions overflowed the dynamic stack into the neighboring
    // wave's scratch and, once raised, the per-queue scratch reservation
    // capped how many HIP queues could dispatch concurrently.
    HIP_CHECK(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
    // Blocking-sync event: hipStreamSynchronize busy-spins; with several
    // engine instances per device that burns cores the HTTP reactors and
    // nghttp2 clients need.  hipEventSynchronize on a hipEventBlockingSync
    // event parks the thread instead, at equal measured throughput.
    // (Multi-instance device-span inflation — 9.3/25/69 ms for the same
    // work at 1/2/4 instances — is unchanged by this, so that cost is in
    // the submission path / device-side interleaving, not the sync wait;
    // see profiles/blocking_sync.log.)
    HIP_CHECK(hipEventCreateWithFlags(
        &sync_ev_, hipEventBlockingSync | hipEventDisableTiming));
    // copy stream for the chunked large-batch pipeline (H2D of chunk c+1
    // and D2H of chunk c-1 overlap chunk c's kernels on stream_)
    HIP_CHECK(hipStreamCreateWithFlags(&stream2_, hipStreamNonBlocking));
    HIP_CHECK(hipStreamCreateWithFlags(&stream3_, hipStreamNonBlocking));
    HIP_CHECK(hipEventCreateWithFlags(
        &sync_ev2_, hipEventBlockingSync | hipEventDisableTiming));
    HIP_CHECK(hipEventCreateWithFlags(
        &sync_ev3_, hipEventBlockingSync | hipEventDisableTiming));
    for (int i = 0; i < PIPE_MAX_CHUNKS; ++i) {
      HIP_CHECK(hipEventCreateWithFlags(&ev_h2d_[i], hipEventDisableTiming));
      HIP_CHECK(hipEventCreateWithFlags(&ev_krn_[i], hipEventDisableTiming));
    }

    upload_blob(msg_table, d_msgs_);
    upload_blob(field_table, d_fields_);
    upload_blob(enum_table, d_enums_);
    upload_blob(enum_values, d_enum_vals_);
    upload_blob(tool_table, d_tools_);
    upload_blob(name_blob, d_names_);
    tables_.msgs = (const MsgEntry*)d_msgs_.p;
    tables_.fields = (const FieldEntry*)d_fields_.p;
    tables_.enums = (const EnumEntry*)d_enums_.p;
    tables_.enum_vals = (const EnumValueEntry*)d_enum_vals_.p;
    tables_.tools = (const ToolEntry*)d_tools_.p;
    tables_.names = (const uint8_t*)d_names_.p;
    tables_.n_msgs = n_msgs;
    tables_.n_tools = n_tools;

    d_in_.alloc(cap_in);
    d_resp_.alloc(cap_in);
    d_scratch_.alloc(cap_scratch);
    d_final_.alloc(cap_final);
    size_t offs = (size_t)(max_batch + 1) * sizeof(uint32_t);
    // Packed control block (off3 | aux2 in ONE allocation each side) and
    // packed result block (results | ids): offsets+aux travel as ONE H2D
    // and results+ids as ONE D2H per batch.  The per-copy-op COUNT is the
    // measured multi-instance ceiling (~3.5k batches/s aggregate across
    // any number of engine instances, profiles/contention_r02.md), so
    // fewer, slightly larger copies beat many small ones.
    ctrl_aux_off_ = align64(offs * 3);
    size_t aux_bytes = (size_t)max_batch * sizeof(int32_t) * 2;
    // control block (off3 | aux2) is ZERO-COPY: the kernels read offsets
    // and routing flags straight from pinned host memory through its
    // device pointer — a few latency-hidden reads per wave beat a ~50 us
    // hipMemcpyAsync SUBMISSION (profiles/contention_r02.md: the engine
    // span is ~85% API-submission time, so the op count is the ceiling)
    h_ctrl_.alloc(ctrl_aux_off_ + aux_bytes);
    void* ctrl_dev = nullptr;
    HIP_CHECK(hipHostGetDevicePointer(&ctrl_dev, h_ctrl_.p, 0));
    d_off3_ = {ctrl_dev, offs * 3};
    d_aux2_ = {(uint8_t*)ctrl_dev + ctrl_aux_off_, aux_bytes};
    h_off_ = {h_ctrl_.p, offs * 3};
    h_aux_ = {(uint8_t*)h_ctrl_.p + ctrl_aux_off_, aux_bytes};
    // results | ids | pb wire in ONE allocation each side: encode returns
    // everything in ONE D2H span (ids sit right after the USED results at
    // rid_off_, set per encode; pb at the fixed rid_cap_ — the dead gap
    // it spans is bandwidth, not an extra submission)
    size_t res_cap = align64((size_t)max_batch * sizeof(SlotResult));
    size_t id_cap = (size_t)max_batch * ID_SLOT_BYTES;
    rid_off_ = res_cap;
    rid_cap_ = align64(res_cap + id_cap);
    d_rid_.alloc(rid_cap_ + cap_pb);
    d_results_ = {d_rid_.p, (size_t)max_batch * sizeof(SlotResult)};
    d_id_slots_ = {(uint8_t*)d_rid_.p + rid_off_, id_cap};
    d_pb_ = {(uint8_t*)d_rid_.p + rid_cap_, cap_pb};
    d_dec_results_.alloc((size_t)max_batch * sizeof(DecodeResult));

    h_in_.alloc(cap_in);
    h_resp_.alloc(cap_in);
    h_final_.alloc(cap_final);
    h_dec_results_.alloc((size_t)max_batch * sizeof(DecodeResult));
    // JSON-RPC id tokens on the host: the native span executor assembles
    // error envelopes in C++ and needs the ids the encode kernel captured
    h_rid_.alloc(rid_cap_ + cap_pb);
    h_results_ = {h_rid_.p, (size_t)max_batch * sizeof(SlotResult)};
    h_id_ = {(uint8_t*)h_rid_.p + rid_off_, id_cap};
    h_pb_ = {(uint8_t*)h_rid_.p + rid_cap_, cap_pb};
  }

  ~Engine() {
    (void)hipStreamSynchronize(stream_);
    (void)hipStreamSynchronize(stream2_);
    (void)hipStreamSynchronize(stream3_);
    (void)hipEventDestroy(sync_ev_);
    (void)hipEventDestroy(sync_ev2_);
    (void)hipEventDestroy(sync_ev3_);
    for (int i = 0; i < PIPE_MAX_CHUNKS; ++i) {
      (void)hipEventDestroy(ev_h2d_[i]);
      (void)hipEventDestroy(ev_krn_[i]);
    }
    (void)hipStreamDestroy(stream_);
    (void)hipStreamDestroy(stream2_);
    (void)hipStreamDestroy(stream3_);
  }

  // park-the-thread stream wait (see sync_ev_ above)
  void sync_stream() {
    HIP_CHECK(hipEventRecord(sync_ev_, stream_));
    HIP_CHECK(hipEventSynchronize(sync_ev_));
  }

  void sync_stream2() {
    HIP_CHECK(hipEventRecord(sync_ev2_, stream2_));
    HIP_CHECK(hipEventSynchronize(sync_ev2_));
  }
  void sync_stream3() {
    HIP_CHECK(hipEventRecord(sync_ev3_, stream3_));
    HIP_CHECK(hipEventSynchronize(sync_ev3_));
  }

  // ---- chunked copy/compute pipeline (large batches only) -----------------
  // Wide-payload batches (wide64: 64 x 64 KB) spend ~1.1 ms/batch in
  // serial H2D + D2H around the kernels.  Splitting the item range into
  // byte-balanced chunks lets chunk c+1's H2D and chunk c-1's D2H run on
  // stream2_ while chunk c's kernels run on stream_.  All per-item kernel
  // arrays index absolutely through the offset tables, so a sub-range
  // launch is just sliced pointers.  Gated OFF for small batches: extra
  // HIP submissions are the known multi-instance contention point
  // (profiles/streams_sweep.log), so serving-sized batches keep the
  // single-shot path.
  static int pipe_chunks() {
    // DEFAULT 1 (off).  Measured negative result (tools/pipe_probe.py,
    // profiles/pipe_chunks_r02.md): the wg kernels are LATENCY-bound —
    // a wide64 batch is 64 workgroups on 256 CUs, so one launch costs one
    // request's serial depth (~2.4 ms) regardless of item count, and C
    // sequential sub-launches cost C x that serial depth (~1.9 ms per
    // extra chunk, single engine, independent of GPU_MAX_HW_QUEUES) —
    // far more than the ~1 ms of copies the chunks overlap.  Kept
    // env-gated for experiments; correctness held (byte-identical wire).
    const char* e = getenv("GGRMCP_PIPE_CHUNKS");  // per-call, like wg_*_min
    int x = e ? atoi(e) : 1;
    return x < 1 ? 1 : (x > PIPE_MAX_CHUNKS ? PIPE_MAX_CHUNKS : x);
  }
  static size_t pipe_min_bytes() {
    // default 1 MiB: serving-shaped batches (<=0.5 MB) stay single-shot —
    // extra submissions are the multi-instance contention point — while
    // wide-payload batches (wide64: ~4 MB each way) always chunk
    const char* e = getenv("GGRMCP_PIPE_MIN");
    return e ? (size_t)atoll(e) : (size_t)(1u << 20);
  }
  // cut [0,n) into <=C ranges balanced by in_off bytes; returns #chunks
  static int pipe_cuts(const uint32_t* off, int n, int C, int* cut) {
    cut[0] = 0;
    int k = 0, lo = 0;
    uint64_t base = off[0], total = off[n] - off[0];
    for (int c = 0; c < C && lo < n; ++c) {
      int hi = (c == C - 1) ? n : lo + 1;
      uint64_t target = base + total * (uint64_t)(c + 1) / (uint64_t)C;
      while (hi < n && off[hi] < target) ++hi;
      cut[++k] = hi;
      lo = hi;
    }
    return k;
  }

  // ---- encode: JSON(-RPC) -> protobuf ------------------------------------
  // mode 0: full envelope; mode 1: bare message.
  // Returns (results u8[n*32] structured, pb memoryview into pinned mem).

  py::tuple encode(py::buffer data, py::array_t<uint32_t> in_off,
                   py::array_t<uint32_t> pb_off, py::object msg_idx, int mode,
                   uint32_t max_depth, uint32_t max_string, uint32_t max_args,
                   int enforce) {
    py::buffer_info din = data.request();
    auto in_off_v = in_off.unchecked<1>();
    auto pb_off_v = pb_off.unchecked<1>();
    int n = (int)in_off_v.shape(0) - 1;
    if (n < 0 || n > max_batch_) throw std::runtime_error("bad batch size");
    size_t in_bytes = in_off_v(n);
    size_t pb_bytes = pb_off_v(n);
    if (in_bytes > h_in_.n || (size_t)din.size < in_bytes)
      throw std::runtime_error("input exceeds engine cap_in");
    if (pb_bytes > d_pb_.n) throw std::runtime_error("pb cap exceeded");
    std::memcpy(h_in_.p, din.ptr, in_bytes);
    uint32_t* h_off = (uint32_t*)h_off_.p;
    std::memcpy(h_off, in_off.data(), (n + 1) * sizeof(uint32_t));
    std::memcpy(h_off + (n + 1), pb_off.data(), (n + 1) * sizeof(uint32_t));
    bool has_idx = stage_msg_idx(msg_idx, n);
    Limits lim{max_depth, max_string, max_args, (uint32_t)enforce};
    return run_encode(n, in_bytes, pb_bytes, has_idx, lim, mode);
  }

  py::tuple encode_list(py::sequence items, py::object msg_idx, int mode,
                        uint32_t max_depth, uint32_t max_string,
                        uint32_t max_args, int enforce) {
    int n = (int)py::len(items);
    if (n < 0 || n > max_batch_) throw std::runtime_error("bad batch size");
    uint32_t* in_off = (uint32_t*)h_off_.p;
    uint32_t* pb_off = in_off + (n + 1);
    size_t acc = 0, pacc = 0;
    uint8_t* dst = (uint8_t*)h_in_.p;
    for (int i = 0; i < n; ++i) {
      const char* ptr;
      size_t len;
      if (!view_of(items[i], &ptr, &len))
        throw std::runtime_error("encode_list: unsupported element type");
      if (acc + len > h_in_.n) throw std::runtime_error("input exceeds cap_in");
      in_off[i] = (uint32_t)acc;
      pb_off[i] = (uint32_t)pacc;
      if (len) std::memcpy(dst + acc, ptr, len);
      acc += len;
      pacc += pb_cap(len);
    }
    in_off[n] = (uint32_t)acc;
    pb_off[n] = (uint32_t)pacc;
    if (pacc > d_pb_.n) throw std::runtime_error("pb cap exceeded");
    bool has_idx = stage_msg_idx(msg_idx, n);
    Limits lim{max_depth, max_string, max_args, (uint32_t)enforce};
    return run_encode(n, acc, pacc, has_idx, lim, mode);
  }

  // ---- decode: protobuf -> JSON(-RPC response) ---------------------------
  // mode 0: envelope using the id slots of the LAST encode batch
  // (slot-aligned); mode 1: bare JSON.

  py::tuple decode(py::buffer data, py::array_t<uint32_t> resp_off,
                   py::array_t<uint32_t> scratch_off,
                   py::array_t<uint32_t> final_off, py::array_t<int32_t> msg_idx,
                   py::object skip, int mode) {
    py::buffer_info din = data.request();
    auto resp_off_v = resp_off.unchecked<1>();
    int n = (int)resp_off_v.shape(0) - 1;
    if (n < 0 || n > max_batch_) throw std::runtime_error("bad batch size");
    size_t resp_bytes = resp_off_v(n);
    auto scratch_off_v = scratch_off.unchecked<1>();
    auto final_off_v = final_off.unchecked<1>();
    size_t scratch_bytes = scratch_off_v(n);
    size_t final_bytes = final_off_v(n);
    if (resp_bytes > h_resp_.n || (size_t)din.size < resp_bytes)
      throw std::runtime_error("input exceeds engine cap_in");
    if (scratch_bytes > d_scratch_.n) throw std::runtime_error("scratch cap");
    if (final_bytes > d_final_.n) throw std::runtime_error("final cap");
    if (mode == 0 && n != last_batch_n_)
      throw std::runtime_error("envelope decode batch must match last encode");
    std::memcpy(h_resp_.p, din.ptr, resp_bytes);
    uint32_t* h_off = (uint32_t*)h_off_.p;
    std::memcpy(h_off, resp_off.data(), (n + 1) * sizeof(uint32_t));
    std::memcpy(h_off + (n + 1), final_off.data(), (n + 1) * sizeof(uint32_t));
    std::memcpy(h_off + 2 * (n + 1), scratch_off.data(),
                (n + 1) * sizeof(uint32_t));
    int32_t* h_aux = (int32_t*)h_aux_.p;
    std::memcpy(h_aux, msg_idx.data(), n * sizeof(int32_t));
    bool has_skip = false;
    if (!skip.is_none()) {
      auto skip_arr = skip.cast<py::array_t<int32_t>>();
      std::memcpy(h_aux + n, skip_arr.data(), n * sizeof(int32_t));
      has_skip = true;
    }
    return run_decode(n, resp_bytes, final_bytes, has_skip, mode);
  }

  py::tuple decode_list(py::sequence items, py::array_t<int32_t> msg_idx,
                        py::object skip, int mode) {
    int n = (int)py::len(items);
    if (n < 0 || n > max_batch_) throw std::runtime_error("bad batch size");
    if (mode == 0 && n != last_batch_n_)
      throw std::runtime_error("envelope decode batch must match last encode");
    uint32_t* resp_off = (uint32_t*)h_off_.p;
    uint32_t* final_off = resp_off + (n + 1);
    uint32_t* scratch_off = resp_off + 2 * (n + 1);
    int32_t* h_aux = (int32_t*)h_aux_.p;
    std::memcpy(h_aux, msg_idx.data(), n * sizeof(int32_t));
    bool has_skip = false;
    const int32_t* skip_src = nullptr;
    py::array_t<int32_t> skip_arr;
    if (!skip.is_none()) {
      skip_arr = skip.cast<py::array_t<int32_t>>();
      skip_src = skip_arr.data();
      has_skip = true;
    }
    size_t acc = 0, sacc = 0, facc = 0;
    uint8_t* dst = (uint8_t*)h_resp_.p;
    for (int i = 0; i < n; ++i) {
      const char* ptr;
      size_t len;
      if (!view_of(items[i], &ptr, &len))
        throw std::runtime_error("decode_list: unsupported element type");
      bool skipped = skip_src && skip_src[i];
      if (skipped) len = 0;  // host handles this slot
      if (acc + len > h_resp_.n) throw std::runtime_error("input exceeds cap_in");
      resp_off[i] = (uint32_t)acc;
      scratch_off[i] = (uint32_t)sacc;
      final_off[i] = (uint32_t)facc;
      if (len) std::memcpy(dst + acc, ptr, len);
      acc += len;
      sacc += scratch_cap(len);
      facc += final_cap(len);
      if (has_skip) h_aux[n + i] = skipped ? 1 : 0;
    }
    resp_off[n] = (uint32_t)acc;
    scratch_off[n] = (uint32_t)sacc;
    final_off[n] = (uint32_t)facc;
    if (sacc > d_scratch_.n) throw std::runtime_error("scratch cap");
    if (facc > d_final_.n) throw std::runtime_error("final cap");
    return run_decode(n, acc, facc, has_skip, mode);
  }

  // ---- fully-native span: encode -> gRPC invoke -> decode ----------------
  // The Python pipeline's per-slot loops (grouping, bytes slicing, list
  // staging) cost ~2 ms per 1024-slot batch; here the pb bytes go straight
  // from pinned memory into the wire client and the responses straight back
  // into the decode staging, all under one GIL hold with releases around
  // the GPU and network waits.  Streaming / error slots are reported back
  // for the Python host paths.  Returns:
  //   (enc_results, dec_results, out_view, pb_view, rpc_errors)
  // rpc_errors: list of None | (grpc_status, message) per slot.
  py::tuple process_span(py::sequence bodies, py::object headers,
                         py::object clients_obj, double timeout_s,
                         uint32_t max_depth, uint32_t max_string,
                         uint32_t max_args, int enforce) {
    if (tool_paths_.empty())
      throw std::runtime_error("engine built without tool metadata");
    std::vector<H2GrpcClient*> clients;
    for (auto el : clients_obj.cast<py::sequence>())
      clients.push_back(el.cast<H2GrpcClient*>());
    if (clients.empty()) throw std::runtime_error("no clients");

    // stage + encode (run_encode manages its own GIL release)
    int n = (int)py::len(bodies);
    if (n < 0 || n > max_batch_) throw std::runtime_error("bad batch size");
    uint32_t* in_off = (uint32_t*)h_off_.p;
    uint32_t* pb_off = in_off + (n + 1);
    size_t acc = 0, pacc = 0;
    uint8_t* dst = (uint8_t*)h_in_.p;
    std::vector<const char*> bptr(n);
    for (int i = 0; i < n; ++i) {
      size_t len;
      if (!view_of(bodies[i], &bptr[i], &len))
        throw std::runtime_error("process_span: unsupported element type");
      if (acc + len > h_in_.n) throw std::runtime_error("input exceeds cap_in");
      in_off[i] = (uint32_t)acc;
      pb_off[i] = (uint32_t)pacc;
      acc += len;
      pacc += pb_cap(len);
    }
    in_off[n] = (uint32_t)acc;
    pb_off[n] = (uint32_t)pacc;
    if (pacc > d_pb_.n) throw std::runtime_error("pb cap exceeded");
    // copies fan out across the staging pool for big batches (the byte
    // buffers stay alive via the borrowed sequence refs; the caller
    // keeps the GIL while the helpers copy)
    stage_copies(in_off, n, acc, [&](int i) {
      size_t len = in_off[i + 1] - in_off[i];
      if (len) std::memcpy(dst + in_off[i], bptr[i], len);
    });
    // per-slot metadata (headers) staged before dropping the GIL
    std::vector<std::vector<std::pair<std::string, std::string>>> metas;
    bool have_headers = !headers.is_none();
    if (have_headers) {
      metas.resize(n);
      auto hseq = headers.cast<py::sequence>();
      for (int i = 0; i < n; ++i) {
        py::object h = hseq[i];
        if (h.is_none()) continue;
        for (auto kv : h.cast<py::dict>())
          metas[i].emplace_back(kv.first.cast<std::string>(),
                                kv.second.cast<std::string>());
      }
    }
    Limits lim{max_depth, max_string, max_args, (uint32_t)enforce};
    // per-stage wall times for /metrics (parse/validate/encode
    // live in run_encode; decode covers pb->JSON + envelope).  Read back
    // via last_stage_ms() — safe because callers hold the engine lock.
    auto t_enc0 = std::chrono::steady_clock::now();
    py::tuple enc_out = run_encode(n, acc, pacc, false, lim, 0);
    last_enc_ms_ = std::chrono::duration<double, std::milli>(
                       std::chrono::steady_clock::now() - t_enc0)
                       .count();

    // route OK unary slots per backend
    SlotResult* rs = (SlotResult*)h_results_.p;
    size_t nb = clients.size();
    std::vector<std::vector<H2GrpcClient::RawCall>> per_be(nb);
    std::vector<std::vector<int>> slots_be(nb);
    uint8_t* pb = (uint8_t*)h_pb_.p;
    for (int i = 0; i < n; ++i) {
      if (rs[i].status != E_OK || (rs[i].flags & SR_SERVER_STREAMING)) continue;
      int tool = rs[i].tool_idx;
      if (tool < 0 || tool >= (int)tool_paths_.size()) continue;
      size_t be = tool < (int)tool_backend_.size()
                      ? (size_t)tool_backend_[tool] % nb
                      : 0;
      const std::string& path = tool_paths_[tool];
      per_be[be].push_back(H2GrpcClient::RawCall{
          path.data(), path.size(), pb + rs[i].pb_off, rs[i].pb_len,
          have_headers && !metas[i].empty() ? &metas[i] : nullptr});
      slots_be[be].push_back(i);
    }

    // streaming slots: copy their request wire OUT now — run_decode's
    // output compaction reuses h_pb_, which would clobber them before the
    // Python streaming fan-out reads the bytes
    py::list stream_pbs(n);
    for (int i = 0; i < n; ++i) {
      if (rs[i].status == E_OK && (rs[i].flags & SR_SERVER_STREAMING))
        stream_pbs[i] = py::bytes((const char*)pb + rs[i].pb_off, rs[i].pb_len);
      else
        stream_pbs[i] = py::none();
    }

    // invoke (GIL released; backends run concurrently inside the client's
    // connection threads — calls to different clients issue sequentially
    // but each returns only after ITS batch completes, so issue all, then
    // wait... simple path: sequential per backend; multi-backend batches
    // overlap because submission is async and the wait is per-batch)
    std::vector<std::vector<std::tuple<int, std::string, std::string>>> res_be(nb);
    auto t_inv0 = std::chrono::steady_clock::now();
    {
      py::gil_scoped_release rel;
      for (size_t b = 0; b < nb; ++b)
        if (!per_be[b].empty())
          res_be[b] = clients[b]->invoke_raw(per_be[b], timeout_s);
    }
    auto t_dec0 = std::chrono::steady_clock::now();
    last_inv_ms_ =
        std::chrono::duration<double, std::milli>(t_dec0 - t_inv0).count();

    // stage responses for decode
    uint32_t* resp_off = (uint32_t*)h_off_.p;
    uint32_t* final_off = resp_off + (n + 1);
    uint32_t* scratch_off = resp_off + 2 * (n + 1);
    int32_t* h_aux = (int32_t*)h_aux_.p;
    std::vector<const std::string*> resp_ptr(n, nullptr);
    py::list rpc_errors(n);
    for (int i = 0; i < n; ++i) rpc_errors[i] = py::none();
    for (size_t b = 0; b < nb; ++b) {
      for (size_t k = 0; k < slots_be[b].size(); ++k) {
        int i = slots_be[b][k];
        auto& r = res_be[b][k];
        int status = std::get<0>(r);
        const std::string& data = std::get<1>(r);
        size_t plen = 0, poff = 0;
        bool compressed = false, truncated = false;
        if (!data.empty()) {
          if (data.size() >= 5) {
            // gRPC frame: [compressed-flag u8][len u32be][payload]
            compressed = data[0] != 0;
            uint32_t len;
            memcpy(&len, data.data() + 1, 4);
            len = ntohl(len);
            if (data.size() >= 5 + (size_t)len) {
              poff = 5;
              plen = len;
            } else {
              truncated = true;
            }
          } else {
            truncated = true;
          }
        }
        // A stream that closed without a decodable grpc-status trailer is
        // malformed (gRPC requires trailers) — never treat it as OK, even
        // when DATA arrived (engine.cpp:489).
        if (status < 0) {
          std::string msg = std::get<2>(r);
          if (msg.empty()) msg = "stream closed without grpc-status";
          rpc_errors[i] = py::make_tuple(2 /*UNKNOWN*/, py::str(msg));
        } else if (status == 0 && compressed) {
          rpc_errors[i] = py::make_tuple(
              13 /*INTERNAL*/,
              py::str("compressed gRPC response frame not supported"));
        } else if (status == 0 && truncated) {
          rpc_errors[i] =
              py::make_tuple(13 /*INTERNAL*/, py::str("truncated gRPC frame"));
        } else if (status == 0) {
          resp_ptr[i] = &std::get<1>(res_be[b][k]);
          // record the unary payload span via aux arrays below
          rs[i].err_pos = (uint32_t)poff;   // reuse: payload offset
          rs[i].aux = (int32_t)plen;        // reuse: payload length
        } else {
          rpc_errors[i] = py::make_tuple(status, py::str(std::get<2>(r)));
        }
      }
    }
    size_t racc = 0, sacc = 0, facc = 0;
    uint8_t* rdst = (uint8_t*)h_resp_.p;
    for (int i = 0; i < n; ++i) {
      size_t len = 0;
      if (resp_ptr[i]) len = (size_t)rs[i].aux;
      if (racc + len > h_resp_.n) throw std::runtime_error("resp cap");
      resp_off[i] = (uint32_t)racc;
      scratch_off[i] = (uint32_t)sacc;
      final_off[i] = (uint32_t)facc;
      h_aux[i] = (rs[i].status == E_OK && !(rs[i].flags & SR_SERVER_STREAMING) &&
                  tool_out_msg_.size() > (size_t)rs[i].tool_idx)
                     ? tool_out_msg_[rs[i].tool_idx]
                     : 0;
      h_aux[n + i] = resp_ptr[i] ? 0 : 1;  // skip slots with no response
      if (len)
        std::memcpy(rdst + racc, resp_ptr[i]->data() + rs[i].err_pos, len);
      racc += len;
      sacc += scratch_cap(len);
      facc += final_cap(len);
    }
    resp_off[n] = (uint32_t)racc;
    scratch_off[n] = (uint32_t)sacc;
    final_off[n] = (uint32_t)facc;
    if (sacc > d_scratch_.n) throw std::runtime_error("scratch cap");
    if (facc > d_final_.n) throw std::runtime_error("final cap");
    py::tuple dec_out = run_decode(n, racc, facc, true, 0);
    last_dec_ms_ = std::chrono::duration<double, std::milli>(
                       std::chrono::steady_clock::now() - t_dec0)
                       .count();

    // Response wire for slots whose GPU decode failed: the host transcodes
    // these already-received bytes instead of re-invoking the RPC (which
    // would duplicate side effects on non-idempotent methods — VERDICT r1
    // item 2 / ADVICE batch.py:680).  res_be strings are still alive here.
    DecodeResult* dr = (DecodeResult*)h_dec_results_.p;
    py::list resp_wires(n);
    for (int i = 0; i < n; ++i) {
      if (resp_ptr[i] && (dr[i].status != E_OK || dr[i].out_len == 0))
        resp_wires[i] = py::bytes(resp_ptr[i]->data() + rs[i].err_pos,
                                  (size_t)rs[i].aux);
      else
        resp_wires[i] = py::none();
    }

    return py::make_tuple(enc_out[0], dec_out[0], dec_out[1], stream_pbs,
                          rpc_errors, resp_wires);
  }

  // ---- fully-native serving span (span_api.h) -----------------------------
  // The GIL-free twin of process_span for the C++ HTTP frontend: the whole
  // tools/call hot path — encode kernel, gRPC invoke, decode kernel, and
  // the response envelopes — with Python needed only for the slot kinds
  // the span reports back (streaming / non-tools-call / host fallbacks).
  // Replaces the reference's per-request broker end-to-end
  // (handler.go:81-139 + reflection.go:333-391).
  bool run_span(const spanapi::SpanIn& in, spanapi::SpanOut* out,
                std::string* err) override {
    using namespace spanapi;
    try {
      int n = (int)in.n;
      if (n <= 0 || n > max_batch_) {
        *err = "bad batch size";
        return false;
      }
      out->slots.assign(n, SlotOut());
      // stage bodies into pinned memory (offsets serial, copies fanned
      // out across the staging pool for big batches)
      uint32_t* in_off = (uint32_t*)h_off_.p;
      uint32_t* pb_off = in_off + (n + 1);
      size_t acc = 0, pacc = 0;
      uint8_t* dst = (uint8_t*)h_in_.p;
      for (int i = 0; i < n; ++i) {
        size_t len = in.body_lens[i];
        if (acc + len > h_in_.n) {
          *err = "input exceeds cap_in";
          return false;
        }
        in_off[i] = (uint32_t)acc;
        pb_off[i] = (uint32_t)pacc;
        acc += len;
        pacc += pb_cap(len);
      }
      in_off[n] = (uint32_t)acc;
      pb_off[n] = (uint32_t)pacc;
      stage_copies(in_off, n, acc, [&](int i) {
        size_t len = in_off[i + 1] - in_off[i];
        if (len) std::memcpy(dst + in_off[i], in.bodies[i], len);
      });
      if (pacc > d_pb_.n) {
        *err = "pb cap exceeded";
        return false;
      }
      Limits lim{in.max_depth, in.max_string, in.max_args,
                 (uint32_t)in.enforce};
      auto t0 = std::chrono::steady_clock::now();
      run_encode_device(n, acc, pacc, false, lim, 0);
      auto t1 = std::chrono::steady_clock::now();
      out->enc_ms =
          std::chrono::duration<double, std::milli>(t1 - t0).count();
      out->enc_gpu_ms = last_enc_gpu_ms_;

      SlotResult* rs = (SlotResult*)h_results_.p;
      uint8_t* pb = (uint8_t*)h_pb_.p;

      // classify slots + route OK unary calls per backend
      size_t nb = in.n_clients;
      std::vector<std::vector<H2GrpcClient::RawCall>> per_be(nb);
      std::vector<std::vector<int>> slots_be(nb);
      for (int i = 0; i < n; ++i) {
        SlotOut& so = out->slots[i];
        so.tool_idx = rs[i].tool_idx;
        int st = rs[i].status;
        if (st == E_OK && (rs[i].flags & SR_SERVER_STREAMING)) {
          // request wire OUT now: decode compaction reuses h_pb_
          so.kind = K_PY_STREAM;
          so.aux.assign((const char*)pb + rs[i].pb_off, rs[i].pb_len);
          continue;
        }
        if (st == E_OK) {
          int tool = rs[i].tool_idx;
          if (tool < 0 || tool >= (int)tool_paths_.size() || nb == 0) {
            so.kind = K_PY_ENC_FALLBACK;
            continue;
          }
          size_t be = tool < (int)tool_backend_.size()
                          ? (size_t)tool_backend_[tool] % nb
                          : 0;
          const std::string& path = tool_paths_[tool];
          per_be[be].push_back(H2GrpcClient::RawCall{
              path.data(), path.size(), pb + rs[i].pb_off, rs[i].pb_len,
              (in.metas && in.metas[i] && !in.metas[i]->empty())
                  ? in.metas[i]
                  : nullptr});
          slots_be[be].push_back(i);
          so.kind = K_FINAL;  // provisional until decode verdict
          continue;
        }
        if (st == E_NOT_TOOLCALL) {
          so.kind = K_PY_NOT_TOOLCALL;
          continue;
        }
        if (st == E_UNSUPPORTED || st == E_OVERFLOW) {
          so.kind = K_PY_ENC_FALLBACK;
          continue;
        }
        so.kind = K_ERR_FINAL;
        int code;
        const char* msg;
        status_to_rpc(st, &code, &msg);
        build_rpc_error(so.aux, i, code, msg);
      }

      // invoke (pure C++; the h2 clients block until their batch resolves)
      std::vector<std::vector<std::tuple<int, std::string, std::string>>>
          res_be(nb);
      for (size_t b = 0; b < nb; ++b)
        if (!per_be[b].empty())
          res_be[b] = ((H2GrpcClient*)in.clients[b])
                          ->invoke_raw(per_be[b], in.timeout_s);
      auto t2 = std::chrono::steady_clock::now();
      out->inv_ms =
          std::chrono::duration<double, std::milli>(t2 - t1).count();

      // parse + stage responses for decode (frame rules match process_span)
      uint32_t* resp_off = (uint32_t*)h_off_.p;
      uint32_t* final_off = resp_off + (n + 1);
      uint32_t* scratch_off = resp_off + 2 * (n + 1);
      int32_t* h_aux = (int32_t*)h_aux_.p;
      std::vector<const std::string*> resp_ptr(n, nullptr);
      for (size_t b = 0; b < nb; ++b) {
        for (size_t k = 0; k < slots_be[b].size(); ++k) {
          int i = slots_be[b][k];
          auto& r = res_be[b][k];
          int status = std::get<0>(r);
          const std::string& data = std::get<1>(r);
          size_t plen = 0, poff = 0;
          bool compressed = false, truncated = false;
          if (!data.empty()) {
            if (data.size() >= 5) {
              compressed = data[0] != 0;
              uint32_t len;
              memcpy(&len, data.data() + 1, 4);
              len = ntohl(len);
              if (data.size() >= 5 + (size_t)len) {
                poff = 5;
                plen = len;
              } else {
                truncated = true;
              }
            } else {
              truncated = true;
            }
          }
          const char* local_err = nullptr;
          if (status < 0) {
            status = 2;
            local_err = std::get<2>(r).empty()
                            ? "stream closed without grpc-status"
                            : std::get<2>(r).c_str();
          } else if (status == 0 && compressed) {
            status = 13;
            local_err = "compressed gRPC response frame not supported";
          } else if (status == 0 && truncated) {
            status = 13;
            local_err = "truncated gRPC frame";
          }
          if (status == 0) {
            resp_ptr[i] = &std::get<1>(res_be[b][k]);
            rs[i].err_pos = (uint32_t)poff;  // reuse: payload offset
            rs[i].aux = (int32_t)plen;       // reuse: payload length
          } else {
            // gRPC failure -> isError tool result, HTTP 200
            // (handler.go:252-259 semantics)
            SlotOut& so = out->slots[i];
            so.kind = K_ERR_FINAL;
            build_grpc_error(so.aux, i, status,
                             local_err ? local_err
                                       : std::get<2>(r).c_str());
          }
        }
      }
      size_t racc = 0, sacc = 0, facc = 0;
      uint8_t* rdst = (uint8_t*)h_resp_.p;
      for (int i = 0; i < n; ++i) {
        size_t len = resp_ptr[i] ? (size_t)rs[i].aux : 0;
        // a response that would overflow an arena falls back PER SLOT with
        // its delivered wire: failing the whole span here would force the
        // caller to either drop or re-invoke already-invoked slots
        if (resp_ptr[i] &&
            (racc + len > h_resp_.n || sacc + scratch_cap(len) > d_scratch_.n ||
             facc + final_cap(len) > d_final_.n)) {
          SlotOut& so = out->slots[i];
          so.kind = K_PY_DEC_FALLBACK;
          so.aux.assign(resp_ptr[i]->data() + rs[i].err_pos, len);
          resp_ptr[i] = nullptr;
          len = 0;
        }
        resp_off[i] = (uint32_t)racc;
        scratch_off[i] = (uint32_t)sacc;
        final_off[i] = (uint32_t)facc;
        h_aux[i] = (resp_ptr[i] &&
                    tool_out_msg_.size() > (size_t)rs[i].tool_idx)
                       ? tool_out_msg_[rs[i].tool_idx]
                       : 0;
        h_aux[n + i] = resp_ptr[i] ? 0 : 1;  // skip slots with no response
        racc += len;
        sacc += scratch_cap(len);
        facc += final_cap(len);
      }
      resp_off[n] = (uint32_t)racc;
      scratch_off[n] = (uint32_t)sacc;
      final_off[n] = (uint32_t)facc;
      stage_copies(resp_off, n, racc, [&](int i) {
        size_t len = resp_off[i + 1] - resp_off[i];
        if (len)
          std::memcpy(rdst + resp_off[i], resp_ptr[i]->data() + rs[i].err_pos,
                      len);
      });
      run_decode_device(n, racc, facc, true, 0);
      auto t3 = std::chrono::steady_clock::now();
      out->dec_ms =
          std::chrono::duration<double, std::milli>(t3 - t2).count();
      out->dec_gpu_ms = last_dec_gpu_ms_;

      DecodeResult* dr = (DecodeResult*)h_dec_results_.p;
      out->blob = compact_used_ ? (const uint8_t*)h_pb_.p
                                : (const uint8_t*)h_final_.p;
      out->blob_len = compact_used_ ? compact_bytes_ : last_final_bytes_;
      for (int i = 0; i < n; ++i) {
        SlotOut& so = out->slots[i];
        if (so.kind != K_FINAL) continue;
        if (resp_ptr[i] && dr[i].status == E_OK && dr[i].out_len > 0) {
          so.off = dr[i].out_off;
          so.len = dr[i].out_len;
        } else if (resp_ptr[i]) {
          // decode rejected a DELIVERED response: hand the received wire
          // to the host transcoder — never re-invoke
          so.kind = K_PY_DEC_FALLBACK;
          so.aux.assign(resp_ptr[i]->data() + rs[i].err_pos,
                        (size_t)rs[i].aux);
        } else {
          // routed but no response tuple (client size mismatch): internal
          so.kind = K_ERR_FINAL;
          build_rpc_error(so.aux, i, -32603, "internal error");
        }
      }
      return true;
    } catch (const std::exception& e) {
      *err = e.what();
      return false;
    }
  }

  // (encode_ms, invoke_ms, decode_ms) of the LAST process_span call;
  // callers serialize spans per engine, so no further synchronization
  py::tuple last_stage_ms() const {
    return py::make_tuple(last_enc_ms_, last_inv_ms_, last_dec_ms_);
  }

  // device-only spans (copies+kernel+sync, measured inside the released-GIL
  // region) of the last run_encode/run_decode — the wall split above minus
  // these is GIL/thread contention
  py::tuple last_gpu_ms() const {
    return py::make_tuple(last_enc_gpu_ms_, last_dec_gpu_ms_);
  }

  int device() const { return device_; }
  int max_batch() const { return max_batch_; }

 private:
  // JSON-RPC id token of a slot (raw JSON captured by the encode kernel;
  // "null" when absent/unparsed — matches the Python path's best effort)
  void append_id(std::string& out, int slot) {
    const SlotResult* rs = (const SlotResult*)h_results_.p;
    uint32_t len = rs[slot].id_len;
    if ((rs[slot].flags & SR_ID_IS_MISSING) || len == 0 ||
        len > (uint32_t)ID_SLOT_BYTES) {
      out += "null";
      return;
    }
    out.append((const char*)h_id_.p + (size_t)slot * ID_SLOT_BYTES, len);
  }

  // {"jsonrpc":"2.0","id":ID,"error":{"code":C,"message":"M"}}
  void build_rpc_error(std::string& out, int slot, int code, const char* msg) {
    out.clear();
    out += "{\"jsonrpc\":\"2.0\",\"id\":";
    append_id(out, slot);
    out += ",\"error\":{\"code\":";
    out += std::to_string(code);
    out += ",\"message\":\"";
    json_escape_append(out, msg, strlen(msg));
    out += "\"}}";
  }

  // gRPC failure -> isError tool result with HTTP 200 (handler.go:252-259;
  // text format matches engine/batch.py _host_slot)
  void build_grpc_error(std::string& out, int slot, int status,
                        const char* detail) {
    out.clear();
    out += "{\"jsonrpc\":\"2.0\",\"id\":";
    append_id(out, slot);
    out += ",\"result\":{\"content\":[{\"type\":\"text\",\"text\":\"gRPC error ";
    out += grpc_status_name(status);
    out += ": ";
    json_escape_append(out, detail, strlen(detail));
    out += "\"}],\"isError\":true}}";
  }

  // fan per-slot staging copies across the pool for big batches,
  // balanced by the offset table's byte prefix (pipe_cuts)
  template <typename F>
  void stage_copies(const uint32_t* off, int n, size_t total, F&& per_slot) {
    int ht = stage_threads();
    if (ht == 0 || total < stage_min_bytes() || n < 2) {
      for (int i = 0; i < n; ++i) per_slot(i);
      return;
    }
    if (!stage_pool_) stage_pool_.reset(new StagePool(ht));
    int K = ht + 1;
    if (K > n) K = n;
    int cut[17];
    int C = pipe_cuts(off, n, K, cut);
    std::function<void(int)> task = [&](int c) {
      for (int i = cut[c]; i < cut[c + 1]; ++i) per_slot(i);
    };
    stage_pool_->run(C, task);
  }

  bool stage_msg_idx(py::object msg_idx, int n) {
    if (msg_idx.is_none()) return false;
    auto arr = msg_idx.cast<py::array_t<int32_t>>();
    std::memcpy(h_aux_.p, arr.data(), n * sizeof(int32_t));
    return true;
  }

  // Device-only encode (no GIL use, callable from native threads).
  // expects: h_in_ staged, h_off_[0..n]=in_off, h_off_[(n+1)..]=pb_off,
  // h_aux_ = msg_idx when has_idx
  void run_encode_device(int n, size_t in_bytes, size_t pb_bytes, bool has_idx,
                         Limits lim, int mode) {
    uint32_t* h_off = (uint32_t*)h_off_.p;
    // route big mode-0 requests to the workgroup-cooperative encode (one
    // request per workgroup; items stage in the decode scratch arena,
    // which is idle during the encode phase).  h_aux is free in mode 0.
    int n_wg = 0;
    if (mode == 0 && !has_idx && 2 * pb_bytes <= d_scratch_.n) {
      uint32_t thr = wg_enc_min();
      int32_t* esk = (int32_t*)h_aux_.p;
      for (int i = 0; i < n; ++i) {
        uint32_t len = h_off[i + 1] - h_off[i];
        esk[i] = len >= thr ? 2 : 0;
        n_wg += esk[i] ? 1 : 0;
      }
    }
    // pack ids right after the USED results span: one D2H returns both
    rid_off_ = align64((size_t)n * sizeof(SlotResult));
    d_id_slots_.p = (uint8_t*)d_rid_.p + rid_off_;
    h_id_.p = (uint8_t*)h_rid_.p + rid_off_;
    auto g0 = std::chrono::steady_clock::now();
    HIP_CHECK(hipSetDevice(device_));
    // chunked pipeline for big batches (mode 0, no msg_idx): overlap the
    // H2D / D2H slices with the kernels instead of serializing them
    if (mode == 0 && !has_idx && n >= 2 && pipe_chunks() > 1 &&
        in_bytes + pb_bytes >= pipe_min_bytes()) {
      int cut[PIPE_MAX_CHUNKS + 1];
      int C = pipe_cuts(h_off, n, pipe_chunks(), cut);
      if (C > 1) {
        run_encode_chunked(n, C, cut, n_wg, lim);
        last_enc_gpu_ms_ = std::chrono::duration<double, std::milli>(
                               std::chrono::steady_clock::now() - g0)
                               .count();
        last_batch_n_ = n;
        return;
      }
    }
    HIP_CHECK(hipMemcpyAsync(d_in_.p, h_in_.p, in_bytes,
                             hipMemcpyHostToDevice, stream_));
    // offsets + aux are zero-copy: the kernels read them from pinned host
    // memory (d_off3_/d_aux2_ are device pointers INTO h_ctrl_)
    uint32_t* d_off = (uint32_t*)d_off3_.p;  // in_off | pb_off contiguous
    int blocks = (int)cdiv(n, WPB);
    if (blocks > 0) {
      hipLaunchKernelGGL(k_json2pb, dim3(blocks), dim3(WPB * WAVE), 0,
                         stream_, (const uint8_t*)d_in_.p,
                         (const uint32_t*)d_off, (uint8_t*)d_pb_.p,
                         (const uint32_t*)d_off + (n + 1),
                         (SlotResult*)d_results_.p, (uint8_t*)d_id_slots_.p,
                         has_idx ? (const int32_t*)d_aux2_.p : nullptr,
                         tables_, lim, n, mode,
                         n_wg > 0 ? (const int32_t*)d_aux2_.p : nullptr);
      HIP_CHECK(hipGetLastError());
    }
    if (n_wg > 0) {
      hipLaunchKernelGGL(k_json2pb_wg, dim3(n), dim3(WG_ENC_WAVES * WAVE), 0,
                         stream_, (const uint8_t*)d_in_.p,
                         (const uint32_t*)d_off, (uint8_t*)d_pb_.p,
                         (const uint32_t*)d_off + (n + 1),
                         (SlotResult*)d_results_.p, (uint8_t*)d_id_slots_.p,
                         (uint8_t*)d_scratch_.p, tables_, lim, n,
                         (const int32_t*)d_aux2_.p, wg_enc_phases(),
                         mw_scan_on());
      HIP_CHECK(hipGetLastError());
    }
    // results + id tokens + pb wire in ONE packed D2H span (ids sit at
    // rid_off_, pb at the fixed rid_cap_; the gap between used ids and
    // rid_cap_ is dead bandwidth, far cheaper than a second submission).
    // The native span's C++ error envelopes need the ids on the host;
    // mode-0 decode keeps reading the device copies.
    HIP_CHECK(hipMemcpyAsync(h_rid_.p, d_rid_.p, rid_cap_ + pb_bytes,
                             hipMemcpyDeviceToHost, stream_));
    sync_stream();
    // pure device span (copies+kernel+sync, no GIL-reacquire wait):
    // separates real GPU time from thread contention in the wall split
    last_enc_gpu_ms_ = std::chrono::duration<double, std::milli>(
                           std::chrono::steady_clock::now() - g0)
                           .count();
    last_batch_n_ = n;
  }

  // Chunked encode (mode 0, no msg_idx — the serving/wide shapes): chunk
  // c+1's input H2D (stream2_) and chunk c-1's pb/results D2H (stream3_)
  // chunk c's kernels run on stream_.  All kernels stay on ONE stream
  // (serialized — no extra kernel-queue contention); only copies overlap.
  // Per-item addressing is absolute through the offset tables, so a
  // sub-range launch is shifted pointers + a smaller n.
  void run_encode_chunked(int n, int C, const int* cut, int n_wg, Limits lim) {
    uint32_t* h_off = (uint32_t*)h_off_.p;
    const uint32_t* h_pb_off = h_off + (n + 1);
    uint32_t* d_off = (uint32_t*)d_off3_.p;  // zero-copy (pinned host)
    for (int c = 0; c < C; ++c) {
      const int lo = cut[c], hi = cut[c + 1], m = hi - lo;
      const size_t a = h_off[lo], b = h_off[hi];
      if (b > a)
        HIP_CHECK(hipMemcpyAsync((uint8_t*)d_in_.p + a, (uint8_t*)h_in_.p + a,
                                 b - a, hipMemcpyHostToDevice, stream2_));
      HIP_CHECK(hipEventRecord(ev_h2d_[c], stream2_));
      HIP_CHECK(hipStreamWaitEvent(stream_, ev_h2d_[c], 0));
      int blocks = (int)cdiv(m, WPB);
      if (blocks > 0) {
        hipLaunchKernelGGL(
            k_json2pb, dim3(blocks), dim3(WPB * WAVE), 0, stream_,
            (const uint8_t*)d_in_.p, (const uint32_t*)d_off + lo,
            (uint8_t*)d_pb_.p, (const uint32_t*)d_off + (n + 1) + lo,
            (SlotResult*)d_results_.p + lo,
            (uint8_t*)d_id_slots_.p + (size_t)lo * ID_SLOT_BYTES, nullptr,
            tables_, lim, m, 0,
            n_wg > 0 ? (const int32_t*)d_aux2_.p + lo : nullptr);
        HIP_CHECK(hipGetLastError());
      }
      if (n_wg > 0) {
        hipLaunchKernelGGL(
            k_json2pb_wg, dim3(m), dim3(WG_ENC_WAVES * WAVE), 0, stream_,
            (const uint8_t*)d_in_.p, (const uint32_t*)d_off + lo,
            (uint8_t*)d_pb_.p, (const uint32_t*)d_off + (n + 1) + lo,
            (SlotResult*)d_results_.p + lo,
            (uint8_t*)d_id_slots_.p + (size_t)lo * ID_SLOT_BYTES,
            (uint8_t*)d_scratch_.p, tables_, lim, m,
            (const int32_t*)d_aux2_.p + lo, wg_enc_phases(), mw_scan_on());
        HIP_CHECK(hipGetLastError());
      }
      HIP_CHECK(hipEventRecord(ev_krn_[c], stream_));
      HIP_CHECK(hipStreamWaitEvent(stream3_, ev_krn_[c], 0));
      const size_t pa = h_pb_off[lo], pe = h_pb_off[hi];
      if (pe > pa)
        HIP_CHECK(hipMemcpyAsync((uint8_t*)h_pb_.p + pa, (uint8_t*)d_pb_.p + pa,
                                 pe - pa, hipMemcpyDeviceToHost, stream3_));
      HIP_CHECK(hipMemcpyAsync((SlotResult*)h_results_.p + lo,
                               (SlotResult*)d_results_.p + lo,
                               (size_t)m * sizeof(SlotResult),
                               hipMemcpyDeviceToHost, stream3_));
      HIP_CHECK(hipMemcpyAsync(
          (uint8_t*)h_id_.p + (size_t)lo * ID_SLOT_BYTES,
          (uint8_t*)d_id_slots_.p + (size_t)lo * ID_SLOT_BYTES,
          (size_t)m * ID_SLOT_BYTES, hipMemcpyDeviceToHost, stream3_));
    }
    sync_stream();
    sync_stream3();
  }

  py::tuple run_encode(int n, size_t in_bytes, size_t pb_bytes, bool has_idx,
                       Limits lim, int mode) {
    {
      py::gil_scoped_release rel;
      run_encode_device(n, in_bytes, pb_bytes, has_idx, lim, mode);
    }
    py::array_t<uint8_t> results((py::ssize_t)(n * sizeof(SlotResult)));
    std::memcpy(results.mutable_data(), h_results_.p, n * sizeof(SlotResult));
    py::memoryview pb_view = py::memoryview::from_memory(h_pb_.p, pb_bytes);
    return py::make_tuple(results, pb_view);
  }

  // Device-only decode (no GIL use, callable from native threads).
  // expects: h_resp_ staged, h_off_ = resp/final/scratch offsets,
  // h_aux_ = msg_idx (+ skip at offset n when has_skip)
  void run_decode_device(int n, size_t resp_bytes, size_t final_bytes,
                         bool has_skip, int mode) {
    uint32_t* h_off = (uint32_t*)h_off_.p;
    int32_t* h_aux = (int32_t*)h_aux_.p;
    // route big mode-0 responses to the workgroup-cooperative kernel
    // (skip tag 2): the classic kernel skips them, k_pb2json_wg decodes
    // one request per workgroup (common.h WG_DEC_*)
    int n_wg = 0;
    if (mode == 0 && has_skip) {
      uint32_t thr = wg_dec_min();
      for (int i = 0; i < n; ++i) {
        if (h_aux[n + i] == 0 && h_off[i + 1] - h_off[i] >= thr) {
          h_aux[n + i] = 2;
          ++n_wg;
        }
      }
    }
    auto g0 = std::chrono::steady_clock::now();
    HIP_CHECK(hipSetDevice(device_));
    // resp_off | final_off | scratch_off contiguous, one copy; same for
    // msg_idx | skip
    uint32_t* d_off = (uint32_t*)d_off3_.p;
    int32_t* d_aux = (int32_t*)d_aux2_.p;
    // chunked pipeline for big response batches: resp H2D slices on
    // stream2_ overlap the per-chunk kernels on stream_ (see
    // run_encode_chunked; the compact/D2H tail below is shared)
    bool chunked = false;
    if (mode == 0 && has_skip && n >= 2 && pipe_chunks() > 1 &&
        resp_bytes >= pipe_min_bytes()) {
      int cut[PIPE_MAX_CHUNKS + 1];
      int C = pipe_cuts(h_off, n, pipe_chunks(), cut);
      if (C > 1) {
        for (int c = 0; c < C; ++c) {
          const int lo = cut[c], hi = cut[c + 1], m = hi - lo;
          const size_t a = h_off[lo], b = h_off[hi];
          if (b > a)
            HIP_CHECK(hipMemcpyAsync((uint8_t*)d_resp_.p + a,
                                     (uint8_t*)h_resp_.p + a, b - a,
                                     hipMemcpyHostToDevice, stream2_));
          HIP_CHECK(hipEventRecord(ev_h2d_[c], stream2_));
          HIP_CHECK(hipStreamWaitEvent(stream_, ev_h2d_[c], 0));
          int cblocks = (int)cdiv(m, WPB);
          if (cblocks > 0) {
            hipLaunchKernelGGL(
                k_pb2json, dim3(cblocks), dim3(WPB * WAVE), 0, stream_,
                (const uint8_t*)d_resp_.p, (const uint32_t*)d_off + lo,
                (const int32_t*)d_aux + lo,
                (const uint8_t*)d_id_slots_.p + (size_t)lo * ID_SLOT_BYTES,
                (const SlotResult*)d_results_.p + lo, (uint8_t*)d_scratch_.p,
                (const uint32_t*)d_off + 2 * (n + 1) + lo,
                (uint8_t*)d_final_.p, (const uint32_t*)d_off + (n + 1) + lo,
                (DecodeResult*)d_dec_results_.p + lo,
                (const int32_t*)d_aux + n + lo, tables_, m, 0);
            HIP_CHECK(hipGetLastError());
          }
          if (n_wg > 0) {
            hipLaunchKernelGGL(
                k_pb2json_wg, dim3(m), dim3(WG_DEC_WAVES * WAVE), 0, stream_,
                (const uint8_t*)d_resp_.p, (const uint32_t*)d_off + lo,
                (const int32_t*)d_aux + lo,
                (const uint8_t*)d_id_slots_.p + (size_t)lo * ID_SLOT_BYTES,
                (const SlotResult*)d_results_.p + lo, (uint8_t*)d_scratch_.p,
                (const uint32_t*)d_off + 2 * (n + 1) + lo,
                (uint8_t*)d_final_.p, (const uint32_t*)d_off + (n + 1) + lo,
                (DecodeResult*)d_dec_results_.p + lo,
                (const int32_t*)d_aux + n + lo, tables_, m, wg_phases());
            HIP_CHECK(hipGetLastError());
          }
        }
        chunked = true;
      }
    }
    if (!chunked) {
      HIP_CHECK(hipMemcpyAsync(d_resp_.p, h_resp_.p, resp_bytes,
                               hipMemcpyHostToDevice, stream_));
      // offsets + msg_idx/skip are zero-copy (pinned host ctrl block)
    }
    int blocks = chunked ? 0 : (int)cdiv(n, WPB);
    if (blocks > 0) {
      hipLaunchKernelGGL(
          k_pb2json, dim3(blocks), dim3(WPB * WAVE), 0, stream_,
          (const uint8_t*)d_resp_.p, (const uint32_t*)d_off,
          (const int32_t*)d_aux, (const uint8_t*)d_id_slots_.p,
          mode == 0 ? (const SlotResult*)d_results_.p : nullptr,
          (uint8_t*)d_scratch_.p, (const uint32_t*)d_off + 2 * (n + 1),
          (uint8_t*)d_final_.p, (const uint32_t*)d_off + (n + 1),
          (DecodeResult*)d_dec_results_.p,
          has_skip ? (const int32_t*)d_aux + n : nullptr, tables_, n, mode);
      HIP_CHECK(hipGetLastError());
    }
    if (n_wg > 0 && !chunked) {
      hipLaunchKernelGGL(
          k_pb2json_wg, dim3(n), dim3(WG_DEC_WAVES * WAVE), 0, stream_,
          (const uint8_t*)d_resp_.p, (const uint32_t*)d_off,
          (const int32_t*)d_aux, (const uint8_t*)d_id_slots_.p,
          (const SlotResult*)d_results_.p, (uint8_t*)d_scratch_.p,
          (const uint32_t*)d_off + 2 * (n + 1), (uint8_t*)d_final_.p,
          (const uint32_t*)d_off + (n + 1), (DecodeResult*)d_dec_results_.p,
          (const int32_t*)d_aux + n, tables_, n, wg_phases());
      HIP_CHECK(hipGetLastError());
    }
    // Device-side compaction: k_tight_scan rewrites out_off to a packed
    // layout and k_compact_out gathers the used bytes, all before the
    // first sync — no host prefix-sum round-trip, no tight-table H2D.
    uint32_t dst_cap = (uint32_t)std::min(d_pb_.n, h_pb_.n);
    if (n > 0) {
      hipLaunchKernelGGL(k_tight_scan, dim3(1), dim3(256), 0, stream_,
                         (DecodeResult*)d_dec_results_.p, n);
      HIP_CHECK(hipGetLastError());
      hipLaunchKernelGGL(k_compact_out, dim3(n), dim3(256), 0, stream_,
                         (const uint8_t*)d_final_.p,
                         (const uint32_t*)d_off + (n + 1),
                         (const DecodeResult*)d_dec_results_.p,
                         (uint8_t*)d_pb_.p, dst_cap, n);
      HIP_CHECK(hipGetLastError());
    }
    HIP_CHECK(hipMemcpyAsync(h_dec_results_.p, d_dec_results_.p,
                             n * sizeof(DecodeResult), hipMemcpyDeviceToHost,
                             stream_));
    sync_stream();
    DecodeResult* rs = (DecodeResult*)h_dec_results_.p;
    uint64_t acc =
        n > 0 ? (uint64_t)rs[n - 1].out_off + ((rs[n - 1].out_len + 3u) & ~3u)
              : 0;
    if (n > 0 && acc <= dst_cap) {
      if (acc)
        HIP_CHECK(hipMemcpyAsync(h_pb_.p, d_pb_.p, acc,
                                 hipMemcpyDeviceToHost, stream_));
      sync_stream();
      compact_bytes_ = acc;
      compact_used_ = true;
    } else {
      // packed total overflows the staging arena (k_compact_out skipped
      // those writes): restore arena offsets and copy the whole arena
      const uint32_t* final_off = h_off + (n + 1);
      for (int i = 0; i < n; ++i) rs[i].out_off = final_off[i];
      HIP_CHECK(hipMemcpyAsync(h_final_.p, d_final_.p, final_bytes,
                               hipMemcpyDeviceToHost, stream_));
      sync_stream();
      compact_used_ = false;
    }
    last_dec_gpu_ms_ = std::chrono::duration<double, std::milli>(
                           std::chrono::steady_clock::now() - g0)
                           .count();
    last_final_bytes_ = final_bytes;
  }

  py::tuple run_decode(int n, size_t resp_bytes, size_t final_bytes,
                       bool has_skip, int mode) {
    {
      py::gil_scoped_release rel;
      run_decode_device(n, resp_bytes, final_bytes, has_skip, mode);
    }
    py::array_t<uint8_t> results((py::ssize_t)(n * sizeof(DecodeResult)));
    std::memcpy(results.mutable_data(), h_dec_results_.p,
                n * sizeof(DecodeResult));
    py::memoryview out_view =
        compact_used_
            ? py::memoryview::from_memory(h_pb_.p, compact_bytes_)
            : py::memoryview::from_memory(h_final_.p, final_bytes);
    return py::make_tuple(results, out_view);
  }

  void upload_blob(py::bytes b, DeviceBuf& buf) {
    std::string s = b;  // copy (init-time only)
    size_t bytes = s.size() ? s.size() : 1;
    buf.alloc(bytes);
    if (s.size())
      HIP_CHECK(hipMemcpy(buf.p, s.data(), s.size(), hipMemcpyHostToDevice));
  }

  int device_;
  int max_batch_;
  double last_enc_ms_ = 0.0, last_inv_ms_ = 0.0, last_dec_ms_ = 0.0;
  double last_enc_gpu_ms_ = 0.0, last_dec_gpu_ms_ = 0.0;
  int last_batch_n_ = -1;
  bool compact_used_ = false;
  size_t compact_bytes_ = 0;
  size_t last_final_bytes_ = 0;
  hipStream_t stream_;
  hipStream_t stream2_ = nullptr;
  hipStream_t stream3_ = nullptr;
  hipEvent_t sync_ev_ = nullptr;
  hipEvent_t sync_ev2_ = nullptr;
  hipEvent_t sync_ev3_ = nullptr;
  hipEvent_t ev_h2d_[PIPE_MAX_CHUNKS] = {};
  hipEvent_t ev_krn_[PIPE_MAX_CHUNKS] = {};
  Tables tables_{};
  DeviceBuf d_msgs_, d_fields_, d_enums_, d_enum_vals_, d_tools_, d_names_;
  DeviceBuf d_in_, d_resp_, d_scratch_, d_final_;
  DeviceBuf d_rid_, d_dec_results_;
  BufView d_off3_, d_aux2_, d_results_, d_id_slots_, d_pb_;
  PinnedBuf h_in_, h_resp_, h_final_, h_dec_results_, h_ctrl_, h_rid_;
  BufView h_off_, h_aux_, h_results_, h_id_, h_pb_;
  std::unique_ptr<StagePool> stage_pool_;  // lazy (see stage_copies)
  size_t ctrl_aux_off_ = 0;  // aux2 offset inside the ctrl block
  size_t rid_off_ = 0;       // ids offset inside the rid|pb block (per batch)
  size_t rid_cap_ = 0;       // pb offset inside the rid|pb block (fixed)
  std::vector<std::string> tool_paths_;
  std::vector<int32_t> tool_out_msg_;
  std::vector<int32_t> tool_backend_;
};

// debug: expose the device's double-double scaling intermediates so host
// and gfx950 arithmetic can be compared bit-for-bit (wg decode work)
__global__ void k_dd_probe(double d, int prec, uint64_t* out) {
  int e10 = (int)floor(log10(d));
  DD s = dd_scale10(d, (prec - 1) - e10);
  out[0] = __builtin_bit_cast(uint64_t, s.hi);
  out[1] = __builtin_bit_cast(uint64_t, s.lo);
  out[2] = dd_round_half_even(s);
  out[3] = (uint64_t)(int64_t)e10;
  int e2 = 0;
  out[4] = gen_digits(d, prec, &e2);
  out[5] = (uint64_t)(int64_t)e2;
  // raw single-step: p1 = d*1e14, e1 = fma(d,1e14,-p1)
  double p1 = d * 1e14;
  double e1 = fma(d, 1e14, -p1);
  out[6] = __builtin_bit_cast(uint64_t, p1);
  out[7] = __builtin_bit_cast(uint64_t, e1);
}

static int device_count() {
  int n = 0;
  hipError_t e = hipGetDeviceCount(&n);
  if (e != hipSuccess) return 0;
  return n;
}

PYBIND11_MODULE(_jsonproto, m) {
  m.doc() = "MI355X batch JSON<->protobuf transcode engine (gfx950 HIP kernels)";
  m.def("device_count", &device_count);
  m.def("dd_probe", [](double d, int prec) {
    uint64_t* dbuf;
    HIP_CHECK(hipMalloc(&dbuf, 8 * sizeof(uint64_t)));
    hipLaunchKernelGGL(k_dd_probe, dim3(1), dim3(1), 0, nullptr, d, prec, dbuf);
    HIP_CHECK(hipGetLastError());
    uint64_t h[8];
    HIP_CHECK(hipMemcpy(h, dbuf, sizeof(h), hipMemcpyDeviceToHost));
    (void)hipFree(dbuf);
    py::dict r;
    r["hi"] = py::cast(*(double*)&h[0]);
    r["lo"] = py::cast(*(double*)&h[1]);
    r["round"] = (unsigned long long)h[2];
    r["e10"] = (long long)(int64_t)h[3];
    r["digits"] = (unsigned long long)h[4];
    r["gen_e10"] = (long long)(int64_t)h[5];
    r["p1"] = py::cast(*(double*)&h[6]);
    r["e1"] = py::cast(*(double*)&h[7]);
    return r;
  });
  // diagnostic: verify the cross-module cast of an _h2grpc Client works in
  // this process (pybind shares its type registry across extensions when
  // the class typeinfo is default-visibility)
  m.def("probe_client", [](py::object obj) {
    H2GrpcClient* c = obj.cast<H2GrpcClient*>();
    return c->healthy();
  });
  m.attr("ID_SLOT_BYTES") = ID_SLOT_BYTES;
  m.attr("SLOT_RESULT_SIZE") = (int)sizeof(SlotResult);
  m.attr("DECODE_RESULT_SIZE") = (int)sizeof(DecodeResult);
  py::class_<Engine>(m, "Engine")
      .def(py::init<int, py::bytes, py::bytes, py::bytes, py::bytes, py::bytes,
                    py::bytes, int, int, int, size_t, size_t, size_t, size_t,
                    py::object, py::object, py::object>(),
           py::arg("device"), py::arg("msg_table"), py::arg("field_table"),
           py::arg("enum_table"), py::arg("enum_values"), py::arg("tool_table"),
           py::arg("name_blob"), py::arg("n_msgs"), py::arg("n_tools"),
           py::arg("max_batch") = 4096, py::arg("cap_in") = 64u << 20,
           py::arg("cap_pb") = 80u << 20, py::arg("cap_scratch") = 128u << 20,
           py::arg("cap_final") = 160u << 20,
           py::arg("tool_paths") = py::none(),
           py::arg("tool_out_msg") = py::none(),
           py::arg("tool_backend") = py::none())
      .def("encode", &Engine::encode, py::arg("data"), py::arg("in_off"),
           py::arg("pb_off"), py::arg("msg_idx") = py::none(),
           py::arg("mode") = 0, py::arg("max_depth") = 10,
           py::arg("max_string") = 1024, py::arg("max_args") = 1u << 20,
           py::arg("enforce") = 1)
      .def("encode_list", &Engine::encode_list, py::arg("items"),
           py::arg("msg_idx") = py::none(), py::arg("mode") = 0,
           py::arg("max_depth") = 10, py::arg("max_string") = 1024,
           py::arg("max_args") = 1u << 20, py::arg("enforce") = 1)
      .def("decode", &Engine::decode, py::arg("data"), py::arg("resp_off"),
           py::arg("scratch_off"), py::arg("final_off"), py::arg("msg_idx"),
           py::arg("skip") = py::none(), py::arg("mode") = 0)
      .def("process_span", &Engine::process_span, py::arg("bodies"),
           py::arg("headers") = py::none(), py::arg("clients") = py::none(),
           py::arg("timeout_s") = 30.0, py::arg("max_depth") = 10,
           py::arg("max_string") = 1024, py::arg("max_args") = 1u << 20,
           py::arg("enforce") = 1)
      .def("decode_list", &Engine::decode_list, py::arg("items"),
           py::arg("msg_idx"), py::arg("skip") = py::none(), py::arg("mode") = 0)
      .def("last_stage_ms", &Engine::last_stage_ms)
      .def("last_gpu_ms", &Engine::last_gpu_ms)
      .def("span_handle",
           [](Engine& e) { return (uintptr_t)(spanapi::ISpanExecutor*)&e; },
           "opaque ISpanExecutor* for the native HTTP frontend "
           "(span_api.h); the engine must outlive the frontend")
      .def_property_readonly("device", &Engine::device)
      .def_property_readonly("max_batch", &Engine::max_batch);
}
