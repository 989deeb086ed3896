// Shared table layouts, slot records and device helpers for the MI355X
// batch-transcode kernels (gfx950 / CDNA4, wave64).
//
// The table formats mirror ggrmcp_amd/engine/tables.py exactly (packed,
// little-endian); static_asserts below pin the sizes.  These kernels are the
// GPU replacement for the reference gateway's per-request CPU hot path
// (aalobaidi/ggRMCP pkg/server/handler.go:81-139 + pkg/grpc/
// reflection.go:333-391): JSON-RPC envelope parse, tool-call validation,
// JSON->protobuf encode, protobuf->JSON decode, response-envelope assembly.
#pragma once

#ifdef GGRMCP_HOST_SIM
#include "host_shim.h"
#else
#include <hip/hip_runtime.h>
#endif
#include <stdint.h>

// ---------------------------------------------------------------------------
// Table records (mirror engine/tables.py; packed little-endian)
// ---------------------------------------------------------------------------

struct __attribute__((packed)) FieldEntry {
  uint64_t hash_json;   // FNV-1a64 of json_name
  uint64_t hash_orig;   // FNV-1a64 of proto field name
  uint32_t number;      // field number
  uint32_t name_off;    // proto name offset in name_blob
  uint32_t json_off;    // json name offset in name_blob
  uint16_t name_len;
  uint16_t json_len;
  int32_t sub_index;    // message index / enum index / -1
  uint8_t kind;         // google.protobuf FieldDescriptor TYPE_*
  uint8_t flags;        // F_* below
  uint8_t oneof_id;     // 255 = none
  uint8_t pad;
};
static_assert(sizeof(FieldEntry) == 40, "FieldEntry layout");

struct __attribute__((packed)) MsgEntry {
  int32_t field_start;
  int32_t field_count;
  int32_t wkt_kind;  // WKT_* below
  int32_t flags;
};
static_assert(sizeof(MsgEntry) == 16, "MsgEntry layout");

struct __attribute__((packed)) EnumEntry {
  int32_t val_start;
  int32_t val_count;
};
static_assert(sizeof(EnumEntry) == 8, "EnumEntry layout");

struct __attribute__((packed)) EnumValueEntry {
  uint64_t hash;
  int32_t number;
  uint32_t name_off;
  uint16_t name_len;
  uint16_t pad;
  int32_t pad2;
};
static_assert(sizeof(EnumValueEntry) == 24, "EnumValueEntry layout");

struct __attribute__((packed)) ToolEntry {
  uint64_t hash;
  int32_t in_msg;
  int32_t out_msg;
  uint32_t name_off;
  uint16_t name_len;
  uint16_t flags;  // 1 = server streaming
};
static_assert(sizeof(ToolEntry) == 24, "ToolEntry layout");

// field flags (tables.py F_*)
enum : uint8_t {
  F_REPEATED = 1,
  F_PACKED = 2,
  F_MAP = 4,
  F_HAS_PRESENCE = 8,
  F_ONEOF = 16,
};

// message well-known-type kinds (tables.py WKT_*)
enum : int32_t {
  WKT_NONE = 0,
  WKT_TIMESTAMP = 1,
  WKT_DURATION = 2,
  WKT_STRUCT = 3,
  WKT_VALUE = 4,
  WKT_LISTVALUE = 5,
  WKT_ANY = 6,
  WKT_FIELDMASK = 7,
  WKT_EMPTY = 8,
  WKT_WRAPPER = 9,
};

// protobuf field kinds (FieldDescriptorProto.Type values)
enum : uint8_t {
  K_DOUBLE = 1, K_FLOAT = 2, K_INT64 = 3, K_UINT64 = 4, K_INT32 = 5,
  K_FIXED64 = 6, K_FIXED32 = 7, K_BOOL = 8, K_STRING = 9, K_GROUP = 10,
  K_MESSAGE = 11, K_BYTES = 12, K_UINT32 = 13, K_ENUM = 14,
  K_SFIXED32 = 15, K_SFIXED64 = 16, K_SINT32 = 17, K_SINT64 = 18,
};

// protobuf wire types
enum : uint32_t { W_VARINT = 0, W_I64 = 1, W_LEN = 2, W_I32 = 5 };

// ---------------------------------------------------------------------------
// Per-request records (kernel <-> host)
// ---------------------------------------------------------------------------

// status codes (mirror engine/batch.py STATUS_*)
enum : int32_t {
  E_OK = 0,
  E_PARSE = 1,           // malformed JSON
  E_INVALID_REQUEST = 2, // bad JSON-RPC envelope
  E_METHOD_NOT_FOUND = 3,// unknown tool name
  E_INVALID_PARAMS = 4,  // schema violation (unknown field, wrong type)
  E_LIMIT = 5,           // depth/string/size limit exceeded
  E_UNSUPPORTED = 6,     // valid but outside the GPU subset -> host fallback
  E_OVERFLOW = 7,        // output buffer cap exceeded -> host fallback
  E_NOT_TOOLCALL = 8,    // well-formed JSON-RPC but method != tools/call
};

// result of the encode (ingest) kernel, one per request slot
struct __attribute__((packed)) SlotResult {
  int32_t status;
  int32_t tool_idx;     // resolved tool (valid when status==E_OK or stream)
  uint32_t pb_off;      // request wire bytes in the pb arena
  uint32_t pb_len;
  uint32_t err_pos;     // input byte position of the error (diagnostics)
  int32_t aux;          // error detail (field number / oneof id / limit kind)
  uint32_t id_len;      // bytes of the JSON-RPC id copied into the id slot
  uint32_t flags;       // SR_* below
};
static_assert(sizeof(SlotResult) == 32, "SlotResult layout");

enum : uint32_t {
  SR_ID_IS_MISSING = 1,   // request had no "id" member
  SR_SERVER_STREAMING = 2,
};

// result of the decode (respond) kernel, one per request slot
struct __attribute__((packed)) DecodeResult {
  int32_t status;
  uint32_t out_off;  // final response bytes in the output arena
  uint32_t out_len;
  uint32_t pad;
};
static_assert(sizeof(DecodeResult) == 16, "DecodeResult layout");

// JSON-RPC id storage per slot (raw JSON token bytes, e.g. `17` or `"abc"`)
constexpr int ID_SLOT_BYTES = 48;

// limits passed to the encode kernel (defaults mirror mcp/validation.py)
struct Limits {
  uint32_t max_depth;       // nesting depth cap (reference: 10)
  uint32_t max_string;      // per-string char cap (reference: 1024)
  uint32_t max_args_bytes;  // whole-arguments byte cap (reference: 1 MB)
  uint32_t enforce;         // 0 = transcode-only mode (no MCP limits)
};

// Table bundle passed to kernels
struct Tables {
  const MsgEntry* msgs;
  const FieldEntry* fields;
  const EnumEntry* enums;
  const EnumValueEntry* enum_vals;
  const ToolEntry* tools;
  const uint8_t* names;
  int32_t n_msgs;
  int32_t n_tools;
};

// ---------------------------------------------------------------------------
// Device helpers
// ---------------------------------------------------------------------------

#ifndef WAVE
#define WAVE 64
#endif
#define DEV __device__ __forceinline__

DEV uint64_t fnv1a64(const uint8_t* p, uint32_t n) {
  uint64_t h = 0xCBF29CE484222325ull;
  for (uint32_t i = 0; i < n; ++i) {
    h ^= p[i];
    h *= 0x100000001B3ull;
  }
  return h;
}

DEV int lane_id() { return threadIdx.x & (WAVE - 1); }

// wave-uniform broadcast of a value held by lane 0
DEV uint32_t bcast0_u32(uint32_t v) { return __shfl(v, 0, WAVE); }
DEV uint64_t bcast0_u64(uint64_t v) { return __shfl(v, 0, WAVE); }

DEV bool is_ws(uint8_t c) { return c == ' ' || c == '\t' || c == '\n' || c == '\r'; }

// ---------------------------------------------------------------------------
// SWAR byte classification over per-lane dwords.  The 1-byte-per-lane
// ballot scans were memory-latency bound (~26 us/KB per wave: one
// dependent 64 B load per iteration); loading 4 bytes per lane amortizes
// the window latency over 4x the bytes.  Returns 0x80 in each byte
// position that matches.
// ---------------------------------------------------------------------------

DEV uint32_t swar_zero(uint32_t w) {
  return (w - 0x01010101u) & ~w & 0x80808080u;
}
DEV uint32_t swar_eq(uint32_t w, uint8_t ch) {
  return swar_zero(w ^ (0x01010101u * ch));
}

// guarded per-lane dword load of s[base..base+4) with `fill` past n
DEV uint32_t load4_or(const uint8_t* s, uint32_t base, uint32_t n,
                      uint8_t fill) {
  if (base + 4 <= n) {
    uint32_t w;
    __builtin_memcpy(&w, s + base, 4);  // unaligned ok on CDNA
    return w;
  }
  uint32_t w = 0;
  for (int j = 0; j < 4; ++j) {
    uint8_t b = base + (uint32_t)j < n ? s[base + j] : fill;
    w |= (uint32_t)b << (8 * j);
  }
  return w;
}

// zigzag
DEV uint64_t zigzag64(int64_t v) { return ((uint64_t)v << 1) ^ (uint64_t)(v >> 63); }
DEV uint32_t zigzag32(int32_t v) { return ((uint32_t)v << 1) ^ (uint32_t)(v >> 31); }
DEV int64_t unzigzag64(uint64_t v) { return (int64_t)(v >> 1) ^ -(int64_t)(v & 1); }

// write a varint; returns byte count (<= 10)
DEV uint32_t put_varint(uint8_t* out, uint64_t v) {
  uint32_t n = 0;
  while (v >= 0x80) {
    out[n++] = (uint8_t)(v | 0x80);
    v >>= 7;
  }
  out[n++] = (uint8_t)v;
  return n;
}

// write v as a NON-MINIMAL varint occupying exactly `width` bytes (protobuf
// parsers accept padded varints).  Used to backfill reserved length slots of
// nested messages / packed arrays without a second sizing pass.
DEV void put_varint_fixed(uint8_t* out, uint64_t v, uint32_t width) {
  for (uint32_t i = 0; i + 1 < width; ++i) {
    out[i] = (uint8_t)(v & 0x7F) | 0x80;
    v >>= 7;
  }
  out[width - 1] = (uint8_t)(v & 0x7F);
}

// read a varint; advances *pos; returns false on truncation/overlong
DEV bool get_varint(const uint8_t* p, uint32_t len, uint32_t* pos, uint64_t* out) {
  uint64_t v = 0;
  uint32_t shift = 0;
  uint32_t i = *pos;
  while (i < len && shift <= 63) {
    uint8_t b = p[i++];
    v |= (uint64_t)(b & 0x7F) << shift;
    if (!(b & 0x80)) {
      *pos = i;
      *out = v;
      return true;
    }
    shift += 7;
  }
  return false;
}

__constant__ double DD_POW10[23] = {
    1e0,  1e1,  1e2,  1e3,  1e4,  1e5,  1e6,  1e7,  1e8,  1e9,  1e10, 1e11,
    1e12, 1e13, 1e14, 1e15, 1e16, 1e17, 1e18, 1e19, 1e20, 1e21, 1e22};

// ---------------------------------------------------------------------------
// double-double arithmetic for near-exact decimal scaling.  A plain
// `d * 10^e` carries ~1 ulp of error per step — enough to generate
// 17-digit blocks that are off by one AND to mis-verify their parse-back
// (found: 301/3 printed as a string that parses to a DIFFERENT double).
// With an fma-compensated (hi, lo) pair the product of an exact power
// step is correct to ~2^-100, so digit generation and round-trip
// verification are exact for every double the kernel handles.
// ---------------------------------------------------------------------------

struct DD {
  double hi, lo;
};

// (hi, lo) * p for EXACT p (a power of ten <= 1e22 (DD_POW10)): Dekker product via fma.
// The fp pragmas are load-bearing: hipcc's device default
// (-ffp-contract=fast-honor-pragmas) reassociated this fma chain and
// applied the compensation term TWICE (measured on gfx950: 301/3 * 1e14
// produced hi one step high and lo = e1 - 2 instead of {p1, e1}).
DEV DD dd_mul_exact(DD a, double p) {
#pragma clang fp contract(off) reassociate(off)
  double p1 = a.hi * p;
  double e1 = fma(a.hi, p, -p1);
  double lo = fma(a.lo, p, e1);
  DD r;
  r.hi = p1 + lo;
  r.lo = (p1 - r.hi) + lo;
  return r;
}

DEV DD dd_div_exact(DD a, double p) {
#pragma clang fp contract(off) reassociate(off)
  double q1 = a.hi / p;
  // residual of the first quotient: a - q1*p, computed exactly
  double r1 = fma(-q1, p, a.hi) + a.lo;
  double q2 = r1 / p;
  DD r;
  r.hi = q1 + q2;
  r.lo = (q1 - r.hi) + q2;
  return r;
}

// collapse a dd-scaled value d*10^e back to one double (correct to
// ~2^-100 through composed steps; inf/zero saturate early)
// POSITIVE input only (the parser applies the sign afterwards); overflow
// saturates to +inf for the caller's range checks
DEV double dd_scale_collapse(double hi, double lo, int e) {
  DD x{hi, lo};
  while (e > 22) {
    x = dd_mul_exact(x, 1e22);
    e -= 22;
    // !(x < bound) also catches the NaN a mid-chain inf cascades into
    if (!(x.hi < 1.7e308)) return HUGE_VAL;
  }
  while (e < -22) {
    x = dd_div_exact(x, 1e22);
    e += 22;
    if (x.hi == 0.0) return 0.0;
  }
  x = e >= 0 ? dd_mul_exact(x, DD_POW10[e]) : dd_div_exact(x, DD_POW10[-e]);
  double r = x.hi + x.lo;
  // at DBL_MAX the renormalization can hit inf with a -inf compensation
  // term, collapsing to NaN for a FINITE true value (found: the exact
  // DBL_MAX decimal encoded as NaN); saturate so callers route the
  // boundary to the host
  if (r != r) return HUGE_VAL;
  return r;
}

// nesting cap for the iterative encode/decode walkers' explicit frame
// stacks (statically-sized private arrays — no dynamic device stack).
// Deeper nesting returns E_LIMIT and transcodes on the host (counted).
// The MCP validation limit is depth 10 (validation.go:163-184), so 16
// leaves headroom for transcode-mode payloads.
constexpr int MAX_RECURSE = 16;

// fixed-width reserved length slots (supports nested payloads < 2^21)
constexpr uint32_t LEN_SLOT = 3;
constexpr uint32_t LEN_SLOT_MAX = (1u << 21) - 1;

// ---------------------------------------------------------------------------
// Workgroup-cooperative decode (k_pb2json_wg): responses at least
// WG_DEC_MIN_BYTES long are decoded one WORKGROUP per request — wave 0
// scans the top-level field runs, then the block's waves decode runs
// concurrently into per-item scratch regions and escape-compact them into
// the final envelope.  Cuts the per-wave ~26 us/KB serialization floor by
// ~the wave count for large payloads (BASELINE config 3).  The engine
// provisions WG_DEC_EXTRA extra scratch per eligible slot (per-item
// regions are padded by WG_DEC_ITEM_PAD for field-name/syntax overhead).
// ---------------------------------------------------------------------------
constexpr uint32_t WG_DEC_MIN_BYTES = 16384;
constexpr int WG_DEC_MAX_ITEMS = 768;
constexpr uint32_t WG_DEC_ITEM_PAD = 256;
constexpr uint32_t WG_DEC_EXTRA =
    WG_DEC_ITEM_PAD * (uint32_t)WG_DEC_MAX_ITEMS + 1024;
constexpr int WG_DEC_WAVES = 16;  // waves per request workgroup

// Workgroup-cooperative ENCODE (k_json2pb_wg): JSON-RPC requests at least
// WG_ENC_MIN_BYTES long split their arguments object into top-level-member
// items (map/array members chunk at entry boundaries — protobuf wire
// concatenation makes chunked runs valid without any joining fix-ups).
// Items encode into a scratch staging area (2x the slot's pb arena span,
// carved from the decode scratch arena, idle during encode) and compact
// into the pb arena.
constexpr uint32_t WG_ENC_MIN_BYTES = 16384;
constexpr int WG_ENC_MAX_ITEMS = 384;
constexpr uint32_t WG_ENC_ITEM_PAD = 160;
constexpr int WG_ENC_WAVES = 16;

// u64 -> decimal text without an addressable temp buffer (a local tmp[20]
// array lands in scratch memory and costs a private-memory round trip per
// digit; emitting MSB-first via the power table keeps everything in VGPRs)
__constant__ uint64_t DEC_P10[20] = {
    1ull, 10ull, 100ull, 1000ull, 10000ull, 100000ull, 1000000ull,
    10000000ull, 100000000ull, 1000000000ull, 10000000000ull,
    100000000000ull, 1000000000000ull, 10000000000000ull,
    100000000000000ull, 1000000000000000ull, 10000000000000000ull,
    100000000000000000ull, 1000000000000000000ull, 10000000000000000000ull};

DEV uint32_t u64_to_dec(uint8_t* out, uint64_t v) {
  uint32_t n = 1;
  while (n < 20 && v >= DEC_P10[n]) ++n;
  for (uint32_t i = 0; i < n; ++i) {
    uint64_t p = DEC_P10[n - 1 - i];
    uint32_t d = (uint32_t)(v / p);
    out[i] = (uint8_t)('0' + d);
    v -= (uint64_t)d * p;
  }
  return n;
}

DEV uint32_t i64_to_dec(uint8_t* out, int64_t v) {
  if (v < 0) {
    out[0] = '-';
    // careful with INT64_MIN
    uint64_t mag = (uint64_t)(~v) + 1ull;
    return 1 + u64_to_dec(out + 1, mag);
  }
  return u64_to_dec(out, (uint64_t)v);
}

// ---------------------------------------------------------------------------
// Strict UTF-8 validation (RFC 3629), wave-parallel: each lane judges its
// own byte.  Lead bytes verify their continuations + overlong/surrogate/
// range rules; continuation bytes verify a lead 1-3 bytes back claims them
// (a bad overlap always also fails the lead's own check).  protojson
// rejects invalid UTF-8 in proto3 strings in BOTH directions (Go proto
// wire unmarshal and protojson.Unmarshal), so the kernels must too:
// decode -> E_UNSUPPORTED (host re-attempt surfaces the error), encode ->
// E_PARSE.  Fuzz-found: wire 0x0A 0x0D 00*12 0x80 decoded to JSON that
// json.loads could not UTF-8-decode.
// ---------------------------------------------------------------------------

DEV bool utf8_byte_ok(const uint8_t* s, uint32_t n, uint32_t i) {
  uint8_t b = s[i];
  if (b < 0x80) return true;
  if (b < 0xC0) {  // continuation: must be claimed by a preceding lead
    if (i >= 1) { uint8_t p = s[i - 1]; if (p >= 0xC2 && p <= 0xF4) return true; }
    if (i >= 2) { uint8_t p = s[i - 2]; if (p >= 0xE0 && p <= 0xF4) return true; }
    if (i >= 3) { uint8_t p = s[i - 3]; if (p >= 0xF0 && p <= 0xF4) return true; }
    return false;
  }
  if (b < 0xC2) return false;  // C0/C1: overlong 2-byte form
  uint8_t c1 = i + 1 < n ? s[i + 1] : 0;
  bool cont1 = (c1 & 0xC0) == 0x80;
  if (b < 0xE0) return cont1;  // 2-byte lead
  uint8_t c2 = i + 2 < n ? s[i + 2] : 0;
  bool cont2 = (c2 & 0xC0) == 0x80;
  if (b < 0xF0) {  // 3-byte lead
    if (!cont1 || !cont2) return false;
    if (b == 0xE0 && c1 < 0xA0) return false;  // overlong
    if (b == 0xED && c1 > 0x9F) return false;  // UTF-16 surrogate range
    return true;
  }
  if (b > 0xF4) return false;  // beyond U+10FFFF
  uint8_t c3 = i + 3 < n ? s[i + 3] : 0;
  if (!cont1 || !cont2 || (c3 & 0xC0) != 0x80) return false;
  if (b == 0xF0 && c1 < 0x90) return false;  // overlong
  if (b == 0xF4 && c1 > 0x8F) return false;  // > U+10FFFF
  return true;
}

// whole-span check; ASCII super-windows (256 B) cost one dword load + one
// ballot per lane — the 1-byte/lane form was memory-latency bound
DEV bool utf8_span_valid(const uint8_t* s, uint32_t n, uint32_t lane) {
  for (uint32_t base = 0; base < n; base += 4u * WAVE) {
    uint32_t off = base + 4u * lane;
    uint32_t w = load4_or(s, off, n, 0);
    if (!__ballot((w & 0x80808080u) != 0)) continue;
    bool ok = true;
    for (uint32_t j = 0; j < 4; ++j) {
      uint32_t i = off + j;
      if (i < n && !utf8_byte_ok(s, n, i)) ok = false;
    }
    if (__ballot(!ok)) return false;
  }
  return true;
}
