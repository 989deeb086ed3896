"""MI355X batch transcode engine.

The GPU replacement for the reference's per-request CPU hot path
(pkg/server/handler.go:81-139 + pkg/grpc/reflection.go:333-391): JSON-RPC
envelope parsing, tool-call validation, JSON -> protobuf wire encoding,
protobuf -> JSON decoding and response-envelope assembly, batched across
concurrent sessions and executed by hand-written gfx950 HIP kernels
(ggrmcp_amd/ops/csrc/).
"""
