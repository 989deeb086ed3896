"""Example proto schemas, mirroring the reference's demo services.

Reference: /root/reference/examples/hello-service/proto/hello.proto (24 lines,
SayHello unary) and proto/complex_service.proto (97 lines: enum, Timestamp,
oneof + map wrapper, recursive Node — 3 services).  Built programmatically
(no protoc in this environment) via ggrmcp_amd.utils.protobuild.
"""

from __future__ import annotations

from ggrmcp_amd.utils.protobuild import FileBuilder


def hello_fdp():
    fb = FileBuilder("hello/hello.proto", "hello")
    (
        fb.message("HelloRequest")
        .field("name", 1, "string")
        .done()
        .message("HelloResponse")
        .field("message", 1, "string")
        .done()
        .service("HelloService")
        .method("SayHello", "HelloRequest", "HelloResponse")
        .done()
    )
    # service HelloService is service index 0 -> path (6,0); its method 0 ->
    # (6,0,2,0). Comments exercise the extraction path (loader.go:195-216).
    fb.comment((6, 0), leading="HelloService greets callers.")
    fb.comment((6, 0, 2, 0), leading="SayHello returns a greeting for the given name.")
    fb.comment((4, 0), leading="The request containing the user's name.")
    return fb.build()


def complex_fdp():
    fb = FileBuilder("complex/complex_service.proto", "complex")
    fb.add_dependency("google/protobuf/timestamp.proto")
    fb.enum("Status", [("STATUS_UNSPECIFIED", 0), ("STATUS_ACTIVE", 1), ("STATUS_INACTIVE", 2)])
    (
        fb.message("UserProfile")
        .field("user_id", 1, "string")
        .field("name", 2, "string")
        .field("status", 3, "enum", enum="Status")
        .field("created_at", 4, "message", message="google.protobuf.Timestamp")
        .field("tags", 5, "string", repeated=True)
        .field("score", 6, "int64")
        .field("rating", 7, "double")
        .field("avatar", 8, "bytes")
        .done()
        .message("GetUserRequest")
        .field("user_id", 1, "string")
        .done()
        .message("Document")
        .field("id", 1, "string")
        .field("text", 2, "string", oneof="content")
        .field("binary", 3, "bytes", oneof="content")
        .map_field("metadata", 4, "string", "string")
        .done()
        .message("Node")
        .field("value", 1, "string")
        .field("children", 2, "message", message="Node", repeated=True)
        .done()
        .message("NodeRequest")
        .field("root", 1, "message", message="Node")
        .field("depth", 2, "uint32")
        .done()
        .service("UserService")
        .method("GetUser", "GetUserRequest", "UserProfile")
        .done()
        .service("DocumentService")
        .method("PutDocument", "Document", "Document")
        .done()
        .service("NodeService")
        .method("Echo", "NodeRequest", "NodeRequest")
        .method("StreamNodes", "NodeRequest", "NodeRequest", server_streaming=True)
        .done()
    )
    return fb.build()


ALL_FDPS = [hello_fdp(), complex_fdp()]
SERVICE_NAMES = [
    "hello.HelloService",
    "complex.UserService",
    "complex.DocumentService",
    "complex.NodeService",
]
