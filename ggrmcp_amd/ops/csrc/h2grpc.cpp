// _h2grpc — python module wrapper around h2grpc_impl.h.

#include "h2grpc_impl.h"

PYBIND11_MODULE(_h2grpc, m) {
  m.doc() = "native gRPC-over-HTTP/2 transport (nghttp2): batch client + bench server";
  py::class_<H2GrpcClient>(m, "Client")
      .def(py::init<const std::string&, int, const std::string&, int, size_t>(),
           py::arg("target"), py::arg("connections") = 4, py::arg("authority") = "",
           py::arg("max_inflight") = 512, py::arg("max_resp_bytes") = 0)
      .def("invoke_stream_batch", &H2GrpcClient::invoke_stream_batch,
           py::arg("paths"), py::arg("payloads"), py::arg("timeout_s") = 30.0,
           py::arg("metadata") = std::vector<std::vector<std::pair<std::string, std::string>>>{})
      .def("invoke_batch", &H2GrpcClient::invoke_batch, py::arg("paths"),
           py::arg("payloads"), py::arg("timeout_s") = 30.0,
           py::arg("metadata") = std::vector<std::vector<std::pair<std::string, std::string>>>{})
      .def("healthy", &H2GrpcClient::healthy)
      .def("raw_handle",
           [](H2GrpcClient& c) { return (uintptr_t)&c; },
           "opaque H2GrpcClient* for the native span executor; the client "
           "must outlive every frontend/engine holding the handle")
      .def("close", &H2GrpcClient::close_all);
  py::class_<H2Server>(m, "Server")
      .def(py::init<const std::string&>(), py::arg("target"))
      .def("add_route", &H2Server::add_route, py::arg("path"), py::arg("kind"))
      .def("request_count", &H2Server::request_count)
      .def("start", &H2Server::start)
      .def("stop", &H2Server::stop);
}
