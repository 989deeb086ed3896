"""ggrmcp-amd: MI355X-native gRPC -> MCP gateway.

A from-scratch re-design of aalobaidi/ggRMCP for AMD Instinct MI355X: the
control plane (gRPC reflection / FileDescriptorSet discovery, JSON-Schema tool
generation, HTTP/JSON-RPC surface, gRPC socket I/O) runs host-side, while the
per-call hot path — JSON-RPC parsing, schema validation and JSON<->protobuf
wire transcoding — is batched across concurrent sessions and executed as
hand-written CDNA4 HIP kernels (gfx950), with sessions sharded data-parallel
across up to 8 GPUs and shared discovery state broadcast via RCCL over xGMI.

See SURVEY.md for the component-by-component map onto the reference.
"""

__version__ = "0.1.0"
