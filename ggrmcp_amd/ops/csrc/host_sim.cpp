// _hostsim — CPU build of the gfx950 kernels (single-lane, see host_shim.h).
//
// Purpose: (1) bit-exact CPU oracle for the device kernels so kernel
// semantics are testable in the GPU-less CI tier; (2) an ASAN/gdb-debuggable
// mirror for chasing kernel memory bugs.  The buffer/offset layout matches
// Engine::encode/decode (engine.cpp) exactly.

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>

#include <cstring>
#include <string>
#include <vector>

#ifndef GGRMCP_HOST_SIM
#error "compile with -DGGRMCP_HOST_SIM -DWAVE=1"
#endif

#include "json2pb.hip"
#include "pb2json.hip"

SimDim3 threadIdx, blockIdx, blockDim, gridDim;

namespace py = pybind11;

namespace {

struct Blob {
  std::string data;
  const uint8_t* ptr() const { return (const uint8_t*)data.data(); }
};

}  // namespace

class HostEngine {
 public:
  HostEngine(py::bytes msg_table, py::bytes field_table, py::bytes enum_table,
             py::bytes enum_values, py::bytes tool_table, py::bytes name_blob,
             int n_msgs, int n_tools)
      : msgs_{msg_table}, fields_{field_table}, enums_{enum_table},
        enum_vals_{enum_values}, tools_{tool_table}, names_{name_blob} {
    t_.msgs = (const MsgEntry*)msgs_.ptr();
    t_.fields = (const FieldEntry*)fields_.ptr();
    t_.enums = (const EnumEntry*)enums_.ptr();
    t_.enum_vals = (const EnumValueEntry*)enum_vals_.ptr();
    t_.tools = (const ToolEntry*)tools_.ptr();
    t_.names = names_.ptr();
    t_.n_msgs = n_msgs;
    t_.n_tools = n_tools;
  }

  py::tuple encode(py::buffer data, py::array_t<uint32_t> in_off,
                   py::array_t<uint32_t> pb_off, py::object msg_idx, int mode,
                   uint32_t max_depth, uint32_t max_string, uint32_t max_args,
                   int enforce) {
    py::buffer_info din = data.request();
    auto in_off_v = in_off.unchecked<1>();
    auto pb_off_v = pb_off.unchecked<1>();
    int n = (int)in_off_v.shape(0) - 1;
    size_t pb_bytes = pb_off_v(n);
    std::vector<uint8_t> pb(pb_bytes, 0);
    std::vector<SlotResult> results(n);
    std::vector<uint8_t> ids((size_t)n * ID_SLOT_BYTES, 0);
    const int32_t* mi = nullptr;
    py::array_t<int32_t> mi_arr;
    if (!msg_idx.is_none()) {
      mi_arr = msg_idx.cast<py::array_t<int32_t>>();
      mi = mi_arr.data();
    }
    Limits lim{max_depth, max_string, max_args, (uint32_t)enforce};
    run_json2pb((const uint8_t*)din.ptr, in_off.data(), pb.data(),
                pb_off.data(), results.data(), ids.data(), mi, n, mode, lim);
    last_results_ = results;
    last_ids_ = ids;
    py::array_t<uint8_t> res({(py::ssize_t)(n * sizeof(SlotResult))});
    std::memcpy(res.mutable_data(), results.data(), n * sizeof(SlotResult));
    return py::make_tuple(res, py::bytes((const char*)pb.data(), pb.size()));
  }

  py::tuple decode(py::buffer data, py::array_t<uint32_t> resp_off,
                   py::array_t<uint32_t> scratch_off,
                   py::array_t<uint32_t> final_off, py::array_t<int32_t> msg_idx,
                   py::object skip, int mode) {
    py::buffer_info din = data.request();
    auto resp_off_v = resp_off.unchecked<1>();
    int n = (int)resp_off_v.shape(0) - 1;
    auto scratch_off_v = scratch_off.unchecked<1>();
    auto final_off_v = final_off.unchecked<1>();
    std::vector<uint8_t> scratch(scratch_off_v(n), 0);
    std::vector<uint8_t> fin(final_off_v(n), 0);
    std::vector<DecodeResult> results(n);
    const int32_t* skip_ptr = nullptr;
    py::array_t<int32_t> skip_arr;
    if (!skip.is_none()) {
      skip_arr = skip.cast<py::array_t<int32_t>>();
      skip_ptr = skip_arr.data();
    }
    run_pb2json((const uint8_t*)din.ptr, resp_off.data(), msg_idx.data(),
                mode == 0 ? last_ids_.data() : nullptr,
                mode == 0 ? last_results_.data() : nullptr, scratch.data(),
                scratch_off.data(), fin.data(), final_off.data(),
                results.data(), skip_ptr, n, mode);
    py::array_t<uint8_t> res({(py::ssize_t)(n * sizeof(DecodeResult))});
    std::memcpy(res.mutable_data(), results.data(), n * sizeof(DecodeResult));
    return py::make_tuple(res, py::bytes((const char*)fin.data(), fin.size()));
  }

 private:
  // drive the kernels one wave (= one request) at a time, WAVE=1 lanes
  void run_json2pb(const uint8_t* in, const uint32_t* in_off, uint8_t* pb,
                   const uint32_t* pb_off, SlotResult* results, uint8_t* ids,
                   const int32_t* mi, int n, int mode, Limits lim) {
    gridDim.x = (unsigned)((n + WPB - 1) / WPB);
    for (unsigned b = 0; b < gridDim.x; ++b) {
      for (int w = 0; w < WPB; ++w) {
        blockIdx.x = b;
        threadIdx.x = (unsigned)(w * WAVE);
        k_json2pb(in, in_off, pb, pb_off, results, ids, mi, t_, lim, n, mode,
                  nullptr);
      }
    }
  }
  void run_pb2json(const uint8_t* resp, const uint32_t* resp_off,
                   const int32_t* msg_idx, const uint8_t* ids,
                   const SlotResult* enc, uint8_t* scratch,
                   const uint32_t* scratch_off, uint8_t* fin,
                   const uint32_t* final_off, DecodeResult* results,
                   const int32_t* skip, int n, int mode) {
    gridDim.x = (unsigned)((n + WPB - 1) / WPB);
    for (unsigned b = 0; b < gridDim.x; ++b) {
      for (int w = 0; w < WPB; ++w) {
        blockIdx.x = b;
        threadIdx.x = (unsigned)(w * WAVE);
        k_pb2json(resp, resp_off, msg_idx, ids, enc, scratch, scratch_off, fin,
                  final_off, results, skip, t_, n, mode);
      }
    }
  }

  Blob msgs_, fields_, enums_, enum_vals_, tools_, names_;
  Tables t_{};
  std::vector<SlotResult> last_results_;
  std::vector<uint8_t> last_ids_;
};

PYBIND11_MODULE(_hostsim, m) {
  m.doc() = "single-lane CPU build of the gfx950 transcode kernels";
  py::class_<HostEngine>(m, "HostEngine")
      .def(py::init<py::bytes, py::bytes, py::bytes, py::bytes, py::bytes,
                    py::bytes, int, int>())
      .def("encode", &HostEngine::encode, py::arg("data"), py::arg("in_off"),
           py::arg("pb_off"), py::arg("msg_idx") = py::none(),
           py::arg("mode") = 0, py::arg("max_depth") = 10,
           py::arg("max_string") = 1024, py::arg("max_args") = 1u << 20,
           py::arg("enforce") = 1)
      .def("decode", &HostEngine::decode, py::arg("data"), py::arg("resp_off"),
           py::arg("scratch_off"), py::arg("final_off"), py::arg("msg_idx"),
           py::arg("skip") = py::none(), py::arg("mode") = 0);
}
