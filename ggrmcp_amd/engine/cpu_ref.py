"""CPU golden transcoder — the differential-test oracle for the GPU kernels.

Uses google.protobuf.json_format (the canonical protojson implementation,
same semantics as the reference's protojson.Unmarshal/Marshal at
reflection.go:351-381).  Also serves as the explicit per-request fallback
for batch slots the kernels flag E_UNSUPPORTED (google.protobuf.Any,
out-of-order wire fields, oversized ids) — counted in engine stats, never
silent.
"""

from __future__ import annotations

from typing import Optional

from google.protobuf import json_format, message_factory
from google.protobuf.descriptor import Descriptor


class CpuTranscoder:
    def __init__(self) -> None:
        self._cls_cache = {}

    def _cls(self, desc: Descriptor):
        cls = self._cls_cache.get(desc.full_name)
        if cls is None:
            cls = message_factory.GetMessageClass(desc)
            self._cls_cache[desc.full_name] = cls
        return cls

    def json_to_pb(self, desc: Descriptor, json_text: str) -> bytes:
        msg = json_format.Parse(json_text or "{}", self._cls(desc)())
        return msg.SerializeToString()

    def pb_to_json(self, desc: Descriptor, wire: bytes) -> str:
        msg = self._cls(desc).FromString(wire)
        return json_format.MessageToJson(
            msg, indent=None, ensure_ascii=False, preserving_proto_field_name=False
        )

    def pb_to_message(self, desc: Descriptor, wire: bytes):
        return self._cls(desc).FromString(wire)
