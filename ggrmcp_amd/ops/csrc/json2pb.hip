// k_json2pb — batched JSON-RPC ingest kernel (gfx950, wave64).
//
// One 64-lane wave per request.  Replaces, in a single launch over the whole
// batch, what the reference gateway does per request on the CPU:
//   * JSON-RPC envelope decode      (handler.go:84, json.Decoder)
//   * envelope + tool-call validate (validation.go:24-61, 96-125)
//   * tool-name resolution          (discovery.go:336-343 atomic map read)
//   * JSON -> protobuf wire encode  (reflection.go:351-357, protojson)
//
// Control flow is wave-uniform (scalar parser state lives in registers and
// is identical across lanes); the 64 lanes act as a data-parallel unit for
// the byte-heavy primitives: whitespace skip, string-end scan, structural
// skip, bulk copies and comparisons (one ballot per 64-byte window instead
// of a per-byte loop).  Nested message / packed-array lengths are written as
// 3-byte NON-MINIMAL varints into reserved slots (protobuf parsers accept
// padded varints), which removes the classic two-pass sizing walk entirely.
//
// Full protojson input semantics for: all scalar kinds (numbers or quoted
// strings, exponent forms), string/bytes (std+URL base64, optional pad),
// bool, enum (name or number), nested messages, repeated (packed emission),
// maps (string/int/bool keys), oneof exclusivity, duplicate-key rejection,
// null handling, and the WKTs Timestamp / Duration / wrappers / Empty /
// Struct / Value / ListValue / FieldMask.  google.protobuf.Any and >48-byte
// request ids return E_UNSUPPORTED so the host transcodes those requests
// (counted, never silent).  Doubles parse with <=1 ulp error outside the
// exact fast path (|exp|<=22, mantissa<2^53).

#include "common.h"

#ifndef DEVN
#ifdef GGRMCP_HOST_SIM
#define DEVN __attribute__((noinline))
#else
#define DEVN __device__ __noinline__
#endif
#endif

#ifndef WPB
#define WPB 4
#endif  // waves (=requests) per 256-thread block

struct Ctx {
  const uint8_t* s;
  uint32_t len, pos;
  uint8_t* out;
  uint32_t opos, ocap;
  Tables t;
  Limits lim;
  int32_t status;
  uint32_t err_pos;
  int32_t aux;
  int lane;
  uint8_t* keybuf;  // 192-byte LDS scratch for escaped keys
};

DEV bool fail(Ctx& c, int32_t code, int32_t aux = 0) {
  if (c.status == E_OK) {
    c.status = code;
    c.err_pos = c.pos;
    c.aux = aux;
  }
  return false;
}

DEV uint8_t peek(Ctx& c) { return c.pos < c.len ? c.s[c.pos] : 0; }

DEV void skip_ws(Ctx& c) {
  while (c.pos < c.len) {
    uint32_t i = c.pos + c.lane;
    uint8_t ch = i < c.len ? c.s[i] : 1;  // sentinel: non-ws
    uint64_t m = __ballot(!is_ws(ch));
    if (m) {
      c.pos += __ffsll((long long)m) - 1;
      return;
    }
    c.pos += WAVE;
  }
}

DEV bool expect(Ctx& c, uint8_t ch) {
  skip_ws(c);
  if (peek(c) != ch) return fail(c, E_PARSE);
  c.pos++;
  return true;
}

// c.pos just after the opening quote; finds the closing quote.
// 4 bytes/lane SWAR windows: the byte stream is latency-bound, so one
// dword load per lane quadruples the bytes covered per round trip.
DEV bool string_end(Ctx& c, uint32_t* end, bool* has_esc) {
  uint32_t p = c.pos;
  bool esc = false;
  while (p < c.len) {
    uint32_t base = p + 4u * (uint32_t)c.lane;
    uint32_t w = load4_or(c.s, base, c.len, '"');
    uint32_t hit = swar_eq(w, '"') | swar_eq(w, '\\');
    uint64_t m = __ballot(hit != 0);
    if (m) {
      int lf = __ffsll((long long)m) - 1;
      uint32_t lh = (uint32_t)__shfl(hit, lf, WAVE);
      uint32_t k = p + 4u * (uint32_t)lf + (uint32_t)(__builtin_ctz(lh) >> 3);
      if (k >= c.len) break;
      if (c.s[k] == '"') {
        *end = k;
        *has_esc = esc;
        return true;
      }
      esc = true;
      p = k + 2;  // skip backslash + escaped char
    } else {
      p += 4u * WAVE;
    }
  }
  c.pos = c.len;
  return fail(c, E_PARSE);
}

// parse a JSON string at pos (expects '"'); returns raw span between quotes
DEV bool string_span(Ctx& c, uint32_t* start, uint32_t* rawlen, bool* has_esc) {
  skip_ws(c);
  if (peek(c) != '"') return fail(c, E_PARSE);
  c.pos++;
  *start = c.pos;
  uint32_t end;
  if (!string_end(c, &end, has_esc)) return false;
  *rawlen = end - *start;
  c.pos = end + 1;
  // JSON text must be valid UTF-8 (protojson.Unmarshal rejects otherwise);
  // escapes are pure ASCII so validating the raw span is exact
  if (!utf8_span_valid(c.s + *start, *rawlen, c.lane))
    return fail(c, E_PARSE);
  return true;
}

DEV uint32_t hex_val(uint8_t ch) {
  if (ch >= '0' && ch <= '9') return ch - '0';
  if (ch >= 'a' && ch <= 'f') return ch - 'a' + 10;
  if (ch >= 'A' && ch <= 'F') return ch - 'A' + 10;
  return 0xFFFFFFFF;
}

// serial unescape of src[0..n) into dst; returns output length or 0xFFFFFFFF
// on malformed escapes.  All lanes compute; only lane 0 stores.
DEV uint32_t unescape_serial(Ctx& c, const uint8_t* src, uint32_t n, uint8_t* dst,
                             uint32_t cap) {
  uint32_t o = 0;
  for (uint32_t i = 0; i < n;) {
    uint8_t ch = src[i];
    if (ch != '\\') {
      if (o >= cap) return 0xFFFFFFFF;
      if (!c.lane) dst[o] = ch;
      o++;
      i++;
      continue;
    }
    if (i + 1 >= n) return 0xFFFFFFFF;
    uint8_t e = src[i + 1];
    i += 2;
    uint8_t dec;
    switch (e) {
      case '"': dec = '"'; break;
      case '\\': dec = '\\'; break;
      case '/': dec = '/'; break;
      case 'b': dec = '\b'; break;
      case 'f': dec = '\f'; break;
      case 'n': dec = '\n'; break;
      case 'r': dec = '\r'; break;
      case 't': dec = '\t'; break;
      case 'u': {
        if (i + 4 > n) return 0xFFFFFFFF;
        uint32_t cp = 0;
        for (int k = 0; k < 4; ++k) {
          uint32_t h = hex_val(src[i + k]);
          if (h == 0xFFFFFFFF) return 0xFFFFFFFF;
          cp = (cp << 4) | h;
        }
        i += 4;
        if (cp >= 0xD800 && cp <= 0xDBFF && i + 6 <= n && src[i] == '\\' &&
            src[i + 1] == 'u') {
          uint32_t lo = 0;
          bool ok = true;
          for (int k = 0; k < 4; ++k) {
            uint32_t h = hex_val(src[i + 2 + k]);
            if (h == 0xFFFFFFFF) { ok = false; break; }
            lo = (lo << 4) | h;
          }
          if (ok && lo >= 0xDC00 && lo <= 0xDFFF) {
            cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
            i += 6;
          }
        }
        // lone/unpaired surrogate: protojson rejects (would encode to
        // invalid UTF-8)
        if (cp >= 0xD800 && cp <= 0xDFFF) return 0xFFFFFFFF;
        // UTF-8 encode
        if (cp < 0x80) {
          if (o + 1 > cap) return 0xFFFFFFFF;
          if (!c.lane) dst[o] = (uint8_t)cp;
          o += 1;
        } else if (cp < 0x800) {
          if (o + 2 > cap) return 0xFFFFFFFF;
          if (!c.lane) {
            dst[o] = 0xC0 | (cp >> 6);
            dst[o + 1] = 0x80 | (cp & 0x3F);
          }
          o += 2;
        } else if (cp < 0x10000) {
          if (o + 3 > cap) return 0xFFFFFFFF;
          if (!c.lane) {
            dst[o] = 0xE0 | (cp >> 12);
            dst[o + 1] = 0x80 | ((cp >> 6) & 0x3F);
            dst[o + 2] = 0x80 | (cp & 0x3F);
          }
          o += 3;
        } else {
          if (o + 4 > cap) return 0xFFFFFFFF;
          if (!c.lane) {
            dst[o] = 0xF0 | (cp >> 18);
            dst[o + 1] = 0x80 | ((cp >> 12) & 0x3F);
            dst[o + 2] = 0x80 | ((cp >> 6) & 0x3F);
            dst[o + 3] = 0x80 | (cp & 0x3F);
          }
          o += 4;
        }
        continue;
      }
      default:
        return 0xFFFFFFFF;
    }
    if (o >= cap) return 0xFFFFFFFF;
    if (!c.lane) dst[o] = dec;
    o++;
  }
  return o;
}

// lane-parallel copy (no escapes)
DEV void wave_copy(Ctx& c, uint8_t* dst, const uint8_t* src, uint32_t n) {
  // dword-per-lane main body (unaligned ok on CDNA); byte tail
  for (uint32_t base = 4u * (uint32_t)c.lane; base + 4 <= n;
       base += 4u * WAVE) {
    uint32_t w;
    __builtin_memcpy(&w, src + base, 4);
    __builtin_memcpy(dst + base, &w, 4);
  }
  for (uint32_t i = (n & ~3u) + c.lane; i < n; i += WAVE) dst[i] = src[i];
}

// lane-parallel byte compare; true if equal
DEV bool wave_equal(Ctx& c, const uint8_t* a, const uint8_t* b, uint32_t n) {
  uint32_t bad = 0;
  for (uint32_t i = c.lane; i < n; i += WAVE) bad |= (a[i] != b[i]);
  return __all(!bad);
}

// ---------------------------------------------------------------------------
// emission helpers (wave-uniform; lane 0 stores)
// ---------------------------------------------------------------------------

DEV uint32_t varint_len(uint64_t v) {
  uint32_t n = 1;
  while (v >= 0x80) {
    v >>= 7;
    ++n;
  }
  return n;
}

DEV bool emit_varint(Ctx& c, uint64_t v) {
  uint32_t n = varint_len(v);
  if (c.opos + n > c.ocap) return fail(c, E_OVERFLOW);
  if (!c.lane) put_varint(c.out + c.opos, v);
  c.opos += n;
  return true;
}

DEV bool emit_tag(Ctx& c, uint32_t number, uint32_t wire) {
  return emit_varint(c, ((uint64_t)number << 3) | wire);
}

DEV bool emit_fixed32(Ctx& c, uint32_t v) {
  if (c.opos + 4 > c.ocap) return fail(c, E_OVERFLOW);
  if (!c.lane) {
    c.out[c.opos] = v & 0xFF;
    c.out[c.opos + 1] = (v >> 8) & 0xFF;
    c.out[c.opos + 2] = (v >> 16) & 0xFF;
    c.out[c.opos + 3] = (v >> 24) & 0xFF;
  }
  c.opos += 4;
  return true;
}

DEV bool emit_fixed64(Ctx& c, uint64_t v) {
  if (c.opos + 8 > c.ocap) return fail(c, E_OVERFLOW);
  if (!c.lane)
    for (int i = 0; i < 8; ++i) c.out[c.opos + i] = (v >> (8 * i)) & 0xFF;
  c.opos += 8;
  return true;
}

// reserve a 3-byte length slot; returns its position
DEV bool reserve_len(Ctx& c, uint32_t* slot) {
  if (c.opos + LEN_SLOT > c.ocap) return fail(c, E_OVERFLOW);
  *slot = c.opos;
  c.opos += LEN_SLOT;
  return true;
}

DEV bool backfill_len(Ctx& c, uint32_t slot) {
  uint32_t n = c.opos - slot - LEN_SLOT;
  if (n > LEN_SLOT_MAX) return fail(c, E_OVERFLOW);
  if (!c.lane) put_varint_fixed(c.out + slot, n, LEN_SLOT);
  return true;
}

// ---------------------------------------------------------------------------
// number parsing (wave-uniform serial; spans are short)
// ---------------------------------------------------------------------------

struct NumVal {
  int cls;  // 0 = integer (neg,mag), 1 = double
  bool neg;
  bool imprecise;  // range undecidable at double precision -> host transcodes
  uint64_t mag;
  double d;
};

__constant__ double POW10[23] = {1e0,  1e1,  1e2,  1e3,  1e4,  1e5,  1e6,  1e7,
                                 1e8,  1e9,  1e10, 1e11, 1e12, 1e13, 1e14, 1e15,
                                 1e16, 1e17, 1e18, 1e19, 1e20, 1e21, 1e22};

// parse decimal text [p, e) -> NumVal.  Returns false if malformed.
DEV bool parse_number_text(const uint8_t* s, uint32_t p, uint32_t e, NumVal* nv) {
  if (p >= e) return false;
  bool neg = false;
  if (s[p] == '-') {
    neg = true;
    ++p;
  } else if (s[p] == '+') {
    ++p;
  }
  if (p >= e) return false;
  uint64_t mag = 0;
  int sig = 0;        // significant digits consumed
  int dec_exp = 0;    // decimal exponent adjustment
  bool overflow = false, any = false, frac = false, expseen = false;
  for (; p < e; ++p) {
    uint8_t ch = s[p];
    if (ch >= '0' && ch <= '9') {
      any = true;
      uint32_t dv = ch - '0';
      // exact u64 accumulation (full 20 digits: int64/uint64 JSON strings
      // use every bit; a 19-digit cap rounded 2^64-nearby values, fuzzer)
      if (!overflow && mag <= (0xFFFFFFFFFFFFFFFFull - dv) / 10) {
        mag = mag * 10 + dv;
        if (mag) sig++;
        if (frac) dec_exp--;
      } else {
        if (!overflow && dv >= 5 && mag != 0xFFFFFFFFFFFFFFFFull)
          ++mag;  // round the kept 19-20 digits by the first dropped one
        overflow = true;
        if (!frac) dec_exp++;
      }
    } else if (ch == '.') {
      if (frac || expseen) return false;
      frac = true;
    } else if (ch == 'e' || ch == 'E') {
      if (!any) return false;
      expseen = true;
      ++p;
      bool eneg = false;
      if (p < e && (s[p] == '-' || s[p] == '+')) {
        eneg = s[p] == '-';
        ++p;
      }
      if (p >= e) return false;
      int ev = 0;
      for (; p < e; ++p) {
        if (s[p] < '0' || s[p] > '9') return false;
        if (ev < 100000) ev = ev * 10 + (s[p] - '0');
      }
      dec_exp += eneg ? -ev : ev;
      break;
    } else {
      return false;
    }
  }
  if (!any) return false;
  nv->imprecise = false;
  if (!frac && !expseen && !overflow && dec_exp == 0) {
    nv->cls = 0;
    nv->neg = neg;
    nv->mag = mag;
    nv->d = (double)mag * (neg ? -1.0 : 1.0);
    return true;
  }
  // double path
  double d = (double)mag;
  if (dec_exp != 0) {
    if (dec_exp > 0 && dec_exp <= 22 && mag < (1ull << 53)) {
      d = d * POW10[dec_exp];
    } else if (dec_exp < 0 && dec_exp >= -22 && mag < (1ull << 53)) {
      d = d / POW10[-dec_exp];
    } else {
      // 54..64-bit mantissas (16-20 significant digits): (double)mag
      // rounds BEFORE the scale, and the second rounding cost 1 ulp
      // (oracle-found: "90.33333333333333" — 9033333333333333 is just
      // above 2^53).  Split mag EXACTLY into 32-bit halves (a signed
      // cast of (mag - trunc) is UB past 2^63) and scale in compensated
      // double-double arithmetic (common.h).  Longer decimals were
      // rounded into mag at accumulation, so this path also covers them
      // to within the 19-digit double-rounding boundary.
      double dh = (double)(uint32_t)(mag >> 32) * 4294967296.0;  // exact
      double dl = (double)(uint32_t)mag;                         // exact
      d = dd_scale_collapse(dh, dl, dec_exp);
      if (isinf(d) && mag != 0) {
        // within a few ulp of DBL_MAX the scaled value can round to inf
        // while strtod stays finite.  Decide by decimal magnitude:
        // value ~= mag * 10^dec_exp.
        int mexp = 0;
        for (uint64_t m = mag; m >= 10; m /= 10) ++mexp;
        int vexp = dec_exp + mexp;
        if (vexp <= 308) nv->imprecise = true;  // boundary -> host transcode
        // vexp >= 309: genuinely out of range, keep inf (caller rejects)
      }
    }
  }
  if (neg) d = -d;
  // correctness boundaries -> host strtod (counted fallback):
  //  * >19-20 significant digits were rounded into mag, so values near
  //    the 19-digit rounding boundary can double-round one ulp off
  //  * the dd scaler loses precision approaching the subnormal range
  if (overflow) nv->imprecise = true;
  {
    double ad = d < 0 ? -d : d;
    if (ad != 0.0 && ad < 1e-306) nv->imprecise = true;
  }
  // integral double that fits -> also expose integer view
  nv->cls = 1;
  nv->neg = neg;
  nv->mag = mag;
  nv->d = d;
  return true;
}

// token end for a bare number/literal
DEV uint32_t token_end(Ctx& c) {
  uint32_t p = c.pos;
  while (p < c.len) {
    uint32_t i = p + c.lane;
    uint8_t ch = i < c.len ? c.s[i] : ',';
    bool tok = (ch >= '0' && ch <= '9') || ch == '+' || ch == '-' || ch == '.' ||
               ch == 'e' || ch == 'E' || (ch >= 'a' && ch <= 'z') ||
               (ch >= 'A' && ch <= 'Z');
    uint64_t m = __ballot(!tok);
    if (m) return p + __ffsll((long long)m) - 1;
    p += WAVE;
  }
  return c.len;
}

// parse a JSON number or quoted number; protojson accepts both for all
// numeric kinds.  Also "Infinity"/"-Infinity"/"NaN" (quoted) for floats.
DEV bool parse_numeric_value(Ctx& c, NumVal* nv, bool allow_nonfinite,
                             int* nonfinite /*0 none, 1 inf, -1 -inf, 2 nan*/) {
  skip_ws(c);
  *nonfinite = 0;
  uint32_t p0, rawlen;
  bool esc;
  if (peek(c) == '"') {
    if (!string_span(c, &p0, &rawlen, &esc)) return false;
    if (esc) return fail(c, E_UNSUPPORTED);
    if (allow_nonfinite) {
      if (rawlen == 8 && c.s[p0] == 'I') {
        *nonfinite = 1;
        return true;
      }
      if (rawlen == 9 && c.s[p0] == '-' && c.s[p0 + 1] == 'I') {
        *nonfinite = -1;
        return true;
      }
      if (rawlen == 3 && c.s[p0] == 'N') {
        *nonfinite = 2;
        return true;
      }
    }
    if (!parse_number_text(c.s, p0, p0 + rawlen, nv)) return fail(c, E_INVALID_PARAMS);
    return true;
  }
  uint32_t e = token_end(c);
  if (e == c.pos) return fail(c, E_PARSE);
  if (!parse_number_text(c.s, c.pos, e, nv)) return fail(c, E_PARSE);
  c.pos = e;
  return true;
}

// integer extraction with range checks; accepts integral doubles (1e3)
DEV bool num_to_i64(const NumVal& nv, int64_t* out) {
  if (nv.cls == 0) {
    if (nv.neg) {
      if (nv.mag > 0x8000000000000000ull) return false;
      *out = (int64_t)(0 - nv.mag);
    } else {
      if (nv.mag > 0x7FFFFFFFFFFFFFFFull) return false;
      *out = (int64_t)nv.mag;
    }
    return true;
  }
  double d = nv.d;
  if (d != trunc(d) || d < -9.223372036854776e18 || d >= 9.223372036854776e18)
    return false;
  *out = (int64_t)d;
  return true;
}

DEV bool num_to_u64(const NumVal& nv, uint64_t* out) {
  if (nv.cls == 0) {
    if (nv.neg && nv.mag) return false;
    *out = nv.mag;
    return true;
  }
  double d = nv.d;
  if (d != trunc(d) || d < 0 || d >= 1.8446744073709552e19) return false;
  *out = (uint64_t)d;
  return true;
}

// ---------------------------------------------------------------------------
// base64 (bytes fields)
// ---------------------------------------------------------------------------

DEV int b64_val(uint8_t ch) {
  if (ch >= 'A' && ch <= 'Z') return ch - 'A';
  if (ch >= 'a' && ch <= 'z') return ch - 'a' + 26;
  if (ch >= '0' && ch <= '9') return ch - '0' + 52;
  if (ch == '+' || ch == '-') return 62;
  if (ch == '/' || ch == '_') return 63;
  return -1;
}

// decode src[0..n) into out (lane0 stores); returns length or 0xFFFFFFFF
DEV uint32_t b64_decode(Ctx& c, const uint8_t* src, uint32_t n, uint8_t* dst,
                        uint32_t cap) {
  while (n && src[n - 1] == '=') --n;
  uint32_t o = 0, acc = 0, nbits = 0;
  for (uint32_t i = 0; i < n; ++i) {
    int v = b64_val(src[i]);
    if (v < 0) return 0xFFFFFFFF;
    acc = (acc << 6) | (uint32_t)v;
    nbits += 6;
    if (nbits >= 8) {
      nbits -= 8;
      if (o >= cap) return 0xFFFFFFFF;
      if (!c.lane) dst[o] = (uint8_t)(acc >> nbits);
      o++;
    }
  }
  return o;
}

// ---------------------------------------------------------------------------
// structural skip
// ---------------------------------------------------------------------------

DEV bool next_structural(Ctx& c, uint32_t* at) {
  uint32_t p = c.pos;
  while (p < c.len) {
    uint32_t base = p + 4u * (uint32_t)c.lane;
    uint32_t w = load4_or(c.s, base, c.len, 0);
    uint32_t hit = swar_eq(w, '{') | swar_eq(w, '}') | swar_eq(w, '[') |
                   swar_eq(w, ']') | swar_eq(w, '"');
    uint64_t m = __ballot(hit != 0);
    if (m) {
      int lf = __ffsll((long long)m) - 1;
      uint32_t lh = (uint32_t)__shfl(hit, lf, WAVE);
      uint32_t k = p + 4u * (uint32_t)lf + (uint32_t)(__builtin_ctz(lh) >> 3);
      if (k >= c.len) break;
      *at = k;
      return true;
    }
    p += 4u * WAVE;
  }
  return fail(c, E_PARSE);
}

// Rolling-window structural scan of a container body: c.pos at the byte
// AFTER the opener; advances to just past the matching closer.  One
// 256 B dword-per-lane load per round trip with in-register serial hit
// processing — the hop-per-token form (next_structural + string_end per
// string) paid one dependent memory round trip per token and dominated
// large-payload scans.  When `marks` is non-null, records the first
// depth-1 comma at or past each `stride`-byte mark (entry-boundary
// chunking for the wg encode scanner).
DEV bool scan_container(Ctx& c, uint32_t* marks, int* n_marks, int cap,
                        uint32_t stride) {
  int depth = 1;
  uint32_t p = c.pos;
  bool in_str = false;
  uint32_t skip_pos = 0xFFFFFFFFu;  // byte escaped by a preceding backslash
  uint32_t next_mark = stride ? c.pos + stride : 0xFFFFFFFFu;
  while (p < c.len) {
    uint32_t off = p + 4u * (uint32_t)c.lane;
    uint32_t w = load4_or(c.s, off, c.len, 0);
    uint32_t hit = swar_eq(w, '"') | swar_eq(w, '\\') | swar_eq(w, '{') |
                   swar_eq(w, '}') | swar_eq(w, '[') | swar_eq(w, ']') |
                   swar_eq(w, ',');
    uint64_t lm = __ballot(hit != 0);
    while (lm) {
      int lf = __ffsll((long long)lm) - 1;
      lm &= lm - 1;
      uint32_t lh = (uint32_t)__shfl(hit, lf, WAVE);
      uint32_t lw = (uint32_t)__shfl(w, lf, WAVE);
      while (lh) {
        uint32_t bidx = (uint32_t)(__builtin_ctz(lh) >> 3);
        lh &= lh - 1;
        uint32_t pos = p + 4u * (uint32_t)lf + bidx;
        if (pos >= c.len) {
          lh = 0;
          break;
        }
        if (pos == skip_pos) continue;
        uint8_t ch = (uint8_t)(lw >> (8 * bidx));
        if (in_str) {
          if (ch == '\\')
            skip_pos = pos + 1;
          else if (ch == '"')
            in_str = false;
        } else if (ch == '"') {
          in_str = true;
        } else if (ch == '{' || ch == '[') {
          ++depth;
        } else if (ch == '}' || ch == ']') {
          if (--depth == 0) {
            c.pos = pos + 1;
            return true;
          }
        } else if (ch == ',' && depth == 1 && pos >= next_mark &&
                   marks && *n_marks < cap) {
          marks[(*n_marks)++] = pos;
          next_mark = pos + stride;
        }
      }
    }
    p += 4u * WAVE;
  }
  return fail(c, E_PARSE);
}

DEV bool skip_value(Ctx& c) {
  skip_ws(c);
  uint8_t ch = peek(c);
  if (ch == '"') {
    uint32_t st, rl;
    bool esc;
    return string_span(c, &st, &rl, &esc);
  }
  if (ch == '{' || ch == '[') {
    c.pos++;
    return scan_container(c, nullptr, nullptr, 0, 0);
  }
  // literal / number
  uint32_t e = token_end(c);
  if (e == c.pos) return fail(c, E_PARSE);
  c.pos = e;
  return true;
}

DEV bool literal_at(Ctx& c, const char* lit, uint32_t n) {
  if (c.pos + n > c.len) return false;
  for (uint32_t i = 0; i < n; ++i)
    if (c.s[c.pos + i] != (uint8_t)lit[i]) return false;
  return true;
}

// ---------------------------------------------------------------------------
// WKT: Timestamp / Duration
// ---------------------------------------------------------------------------

DEV int64_t days_from_civil(int64_t y, int64_t m, int64_t d) {
  y -= m <= 2;
  int64_t era = (y >= 0 ? y : y - 399) / 400;
  int64_t yoe = y - era * 400;
  int64_t doy = (153 * (m + (m > 2 ? -3 : 9)) + 2) / 5 + d - 1;
  int64_t doe = yoe * 365 + yoe / 4 - yoe / 100 + doy;
  return era * 146097 + doe - 719468;
}

DEV bool parse_timestamp(Ctx& c, const uint8_t* s, uint32_t n, int64_t* secs,
                         int32_t* nanos) {
  // YYYY-MM-DDTHH:MM:SS[.fff...][Z|±HH:MM]
  if (n < 20) return false;
  auto dig = [&](uint32_t i) -> int { return s[i] - '0'; };
  for (uint32_t i : {0u, 1u, 2u, 3u, 5u, 6u, 8u, 9u, 11u, 12u, 14u, 15u, 17u, 18u})
    if (s[i] < '0' || s[i] > '9') return false;
  if (s[4] != '-' || s[7] != '-' || (s[10] != 'T' && s[10] != 't') || s[13] != ':' ||
      s[16] != ':')
    return false;
  int64_t y = dig(0) * 1000 + dig(1) * 100 + dig(2) * 10 + dig(3);
  int64_t mo = dig(5) * 10 + dig(6);
  int64_t da = dig(8) * 10 + dig(9);
  int64_t hh = dig(11) * 10 + dig(12);
  int64_t mi = dig(14) * 10 + dig(15);
  int64_t ss = dig(17) * 10 + dig(18);
  uint32_t i = 19;
  int32_t ns = 0;
  if (i < n && s[i] == '.') {
    ++i;
    int scale = 100000000;
    while (i < n && s[i] >= '0' && s[i] <= '9') {
      ns += (s[i] - '0') * scale;
      scale /= 10;
      ++i;
    }
  }
  int64_t tz = 0;
  if (i < n && (s[i] == 'Z' || s[i] == 'z')) {
    ++i;
  } else if (i < n && (s[i] == '+' || s[i] == '-')) {
    if (i + 6 > n || s[i + 3] != ':') return false;
    int64_t th = (s[i + 1] - '0') * 10 + (s[i + 2] - '0');
    int64_t tm = (s[i + 4] - '0') * 10 + (s[i + 5] - '0');
    tz = (th * 3600 + tm * 60) * (s[i] == '+' ? 1 : -1);
    i += 6;
  } else {
    return false;
  }
  if (i != n) return false;
  *secs = days_from_civil(y, mo, da) * 86400 + hh * 3600 + mi * 60 + ss - tz;
  *nanos = ns;
  return true;
}

DEV bool parse_duration(const uint8_t* s, uint32_t n, int64_t* secs, int32_t* nanos) {
  if (n < 2 || s[n - 1] != 's') return false;
  n -= 1;
  uint32_t i = 0;
  bool neg = false;
  if (s[0] == '-') {
    neg = true;
    i = 1;
  }
  int64_t sec = 0;
  bool any = false;
  while (i < n && s[i] >= '0' && s[i] <= '9') {
    sec = sec * 10 + (s[i] - '0');
    ++i;
    any = true;
  }
  int32_t ns = 0;
  if (i < n && s[i] == '.') {
    ++i;
    int scale = 100000000;
    while (i < n && s[i] >= '0' && s[i] <= '9') {
      ns += (s[i] - '0') * scale;
      scale /= 10;
      ++i;
      any = true;
    }
  }
  if (!any || i != n) return false;
  *secs = neg ? -sec : sec;
  *nanos = neg ? -ns : ns;
  return true;
}

// ---------------------------------------------------------------------------
// recursive encoder
// ---------------------------------------------------------------------------

// encode a string field's payload (tag already emitted): LEN slot + bytes
DEV bool encode_string_payload(Ctx& c, uint32_t start, uint32_t rawlen, bool esc) {
  uint32_t slot;
  if (!reserve_len(c, &slot)) return false;
  if (!esc) {
    if (c.opos + rawlen > c.ocap) return fail(c, E_OVERFLOW);
    wave_copy(c, c.out + c.opos, c.s + start, rawlen);
    c.opos += rawlen;
  } else {
    uint32_t n = unescape_serial(c, c.s + start, rawlen, c.out + c.opos,
                                 c.ocap - c.opos);
    if (n == 0xFFFFFFFF) return fail(c, E_PARSE);
    c.opos += n;
  }
  return backfill_len(c, slot);
}


DEVN bool encode_scalar_field(Ctx& c, const FieldEntry& f) {
  switch (f.kind) {
    case K_STRING: {
      uint32_t st, rl;
      bool esc;
      if (!string_span(c, &st, &rl, &esc)) return false;
      if (c.lim.enforce && rl > c.lim.max_string) return fail(c, E_LIMIT, 2);
      if (!emit_tag(c, f.number, W_LEN)) return false;
      return encode_string_payload(c, st, rl, esc);
    }
    case K_BYTES: {
      uint32_t st, rl;
      bool esc;
      if (!string_span(c, &st, &rl, &esc)) return false;
      if (esc) return fail(c, E_UNSUPPORTED);
      if (!emit_tag(c, f.number, W_LEN)) return false;
      uint32_t slot;
      if (!reserve_len(c, &slot)) return false;
      uint32_t n = b64_decode(c, c.s + st, rl, c.out + c.opos, c.ocap - c.opos);
      if (n == 0xFFFFFFFF) return fail(c, E_INVALID_PARAMS, (int)f.number);
      c.opos += n;
      return backfill_len(c, slot);
    }
    case K_BOOL: {
      skip_ws(c);
      bool v;
      if (literal_at(c, "true", 4)) {
        v = true;
        c.pos += 4;
      } else if (literal_at(c, "false", 5)) {
        v = false;
        c.pos += 5;
      } else {
        return fail(c, E_INVALID_PARAMS, (int)f.number);
      }
      if (!emit_tag(c, f.number, W_VARINT)) return false;
      return emit_varint(c, v ? 1 : 0);
    }
    case K_DOUBLE:
    case K_FLOAT: {
      NumVal nv;
      int nonf;
      if (!parse_numeric_value(c, &nv, true, &nonf)) return false;
      double d;
      if (nonf == 1) d = HUGE_VAL;
      else if (nonf == -1) d = -HUGE_VAL;
      else if (nonf == 2) d = nan("");
      else d = nv.cls == 0 ? (nv.neg ? -(double)nv.mag : (double)nv.mag) : nv.d;
      if (nv.imprecise && !nonf) return fail(c, E_UNSUPPORTED, (int)f.number);
      if (f.kind == K_DOUBLE) {
        if (isinf(d) && !nonf)  // finite text out of double range (protojson rejects)
          return fail(c, E_INVALID_PARAMS, (int)f.number);
        if (!emit_tag(c, f.number, W_I64)) return false;
        uint64_t bits = __builtin_bit_cast(uint64_t, d);
        return emit_fixed64(c, bits);
      }
      float fv = (float)d;
      if (isinf(fv) && !isinf(d) && !nonf)  // out of float range
        return fail(c, E_INVALID_PARAMS, (int)f.number);
      if (!emit_tag(c, f.number, W_I32)) return false;
      return emit_fixed32(c, __builtin_bit_cast(uint32_t, fv));
    }
    case K_ENUM: {
      skip_ws(c);
      int32_t number;
      if (peek(c) == '"') {
        uint32_t st, rl;
        bool esc;
        if (!string_span(c, &st, &rl, &esc)) return false;
        if (esc) return fail(c, E_UNSUPPORTED);
        uint64_t h = fnv1a64(c.s + st, rl);
        const EnumEntry& ee = c.t.enums[f.sub_index];
        bool found = false;
        for (int i = 0; i < ee.val_count; ++i) {
          const EnumValueEntry& ev = c.t.enum_vals[ee.val_start + i];
          if (ev.hash == h && ev.name_len == rl &&
              wave_equal(c, c.t.names + ev.name_off, c.s + st, rl)) {
            number = ev.number;
            found = true;
            break;
          }
        }
        if (!found) return fail(c, E_INVALID_PARAMS, (int)f.number);
      } else if (literal_at(c, "null", 4)) {
        // only google.protobuf.NullValue accepts null; callers filter null
        // before reaching here, so treat as 0
        c.pos += 4;
        number = 0;
      } else {
        NumVal nv;
        int nonf;
        if (!parse_numeric_value(c, &nv, false, &nonf)) return false;
        int64_t v;
        if (!num_to_i64(nv, &v) || v < -2147483648ll || v > 2147483647ll)
          return fail(c, E_INVALID_PARAMS, (int)f.number);
        number = (int32_t)v;
      }
      if (!emit_tag(c, f.number, W_VARINT)) return false;
      return emit_varint(c, (uint64_t)(int64_t)number);
    }
    default: {  // integer kinds
      NumVal nv;
      int nonf;
      if (!parse_numeric_value(c, &nv, false, &nonf)) return false;
      uint64_t payload;
      uint32_t wire = W_VARINT;
      switch (f.kind) {
        case K_INT64: {
          int64_t v;
          if (!num_to_i64(nv, &v)) return fail(c, E_INVALID_PARAMS, (int)f.number);
          payload = (uint64_t)v;
          break;
        }
        case K_SINT64: {
          int64_t v;
          if (!num_to_i64(nv, &v)) return fail(c, E_INVALID_PARAMS, (int)f.number);
          payload = zigzag64(v);
          break;
        }
        case K_SFIXED64: {
          int64_t v;
          if (!num_to_i64(nv, &v)) return fail(c, E_INVALID_PARAMS, (int)f.number);
          if (!emit_tag(c, f.number, W_I64)) return false;
          return emit_fixed64(c, (uint64_t)v);
        }
        case K_UINT64: {
          uint64_t v;
          if (!num_to_u64(nv, &v)) return fail(c, E_INVALID_PARAMS, (int)f.number);
          payload = v;
          break;
        }
        case K_FIXED64: {
          uint64_t v;
          if (!num_to_u64(nv, &v)) return fail(c, E_INVALID_PARAMS, (int)f.number);
          if (!emit_tag(c, f.number, W_I64)) return false;
          return emit_fixed64(c, v);
        }
        case K_INT32: {
          int64_t v;
          if (!num_to_i64(nv, &v) || v < -2147483648ll || v > 2147483647ll)
            return fail(c, E_INVALID_PARAMS, (int)f.number);
          payload = (uint64_t)v;  // sign-extended like protobuf int32
          break;
        }
        case K_SINT32: {
          int64_t v;
          if (!num_to_i64(nv, &v) || v < -2147483648ll || v > 2147483647ll)
            return fail(c, E_INVALID_PARAMS, (int)f.number);
          payload = zigzag32((int32_t)v);
          break;
        }
        case K_SFIXED32: {
          int64_t v;
          if (!num_to_i64(nv, &v) || v < -2147483648ll || v > 2147483647ll)
            return fail(c, E_INVALID_PARAMS, (int)f.number);
          if (!emit_tag(c, f.number, W_I32)) return false;
          return emit_fixed32(c, (uint32_t)(int32_t)v);
        }
        case K_UINT32: {
          uint64_t v;
          if (!num_to_u64(nv, &v) || v > 0xFFFFFFFFull)
            return fail(c, E_INVALID_PARAMS, (int)f.number);
          payload = v;
          break;
        }
        case K_FIXED32: {
          uint64_t v;
          if (!num_to_u64(nv, &v) || v > 0xFFFFFFFFull)
            return fail(c, E_INVALID_PARAMS, (int)f.number);
          if (!emit_tag(c, f.number, W_I32)) return false;
          return emit_fixed32(c, (uint32_t)v);
        }
        default:
          return fail(c, E_UNSUPPORTED);
      }
      if (!emit_tag(c, f.number, wire)) return false;
      return emit_varint(c, payload);
    }
  }
}


// packed element (no tag) for numeric repeated fields
DEV bool encode_packed_element(Ctx& c, const FieldEntry& f) {
  NumVal nv;
  int nonf;
  switch (f.kind) {
    case K_BOOL: {
      skip_ws(c);
      if (literal_at(c, "true", 4)) {
        c.pos += 4;
        return emit_varint(c, 1);
      }
      if (literal_at(c, "false", 5)) {
        c.pos += 5;
        return emit_varint(c, 0);
      }
      return fail(c, E_INVALID_PARAMS, (int)f.number);
    }
    case K_DOUBLE:
    case K_FLOAT: {
      if (!parse_numeric_value(c, &nv, true, &nonf)) return false;
      double d;
      if (nonf == 1) d = HUGE_VAL;
      else if (nonf == -1) d = -HUGE_VAL;
      else if (nonf == 2) d = nan("");
      else d = nv.cls == 0 ? (nv.neg ? -(double)nv.mag : (double)nv.mag) : nv.d;
      if (nv.imprecise && !nonf) return fail(c, E_UNSUPPORTED, (int)f.number);
      if (f.kind == K_DOUBLE) return emit_fixed64(c, __builtin_bit_cast(uint64_t, d));
      return emit_fixed32(c, __builtin_bit_cast(uint32_t, (float)d));
    }
    case K_ENUM: {
      skip_ws(c);
      if (peek(c) == '"') {
        uint32_t st, rl;
        bool esc;
        if (!string_span(c, &st, &rl, &esc)) return false;
        if (esc) return fail(c, E_UNSUPPORTED);
        uint64_t h = fnv1a64(c.s + st, rl);
        const EnumEntry& ee = c.t.enums[f.sub_index];
        for (int i = 0; i < ee.val_count; ++i) {
          const EnumValueEntry& ev = c.t.enum_vals[ee.val_start + i];
          if (ev.hash == h && ev.name_len == rl &&
              wave_equal(c, c.t.names + ev.name_off, c.s + st, rl))
            return emit_varint(c, (uint64_t)(int64_t)ev.number);
        }
        return fail(c, E_INVALID_PARAMS, (int)f.number);
      }
      if (!parse_numeric_value(c, &nv, false, &nonf)) return false;
      int64_t v;
      if (!num_to_i64(nv, &v)) return fail(c, E_INVALID_PARAMS, (int)f.number);
      return emit_varint(c, (uint64_t)v);
    }
    default: {
      if (!parse_numeric_value(c, &nv, false, &nonf)) return false;
      switch (f.kind) {
        case K_INT64: {
          int64_t v;
          if (!num_to_i64(nv, &v)) return fail(c, E_INVALID_PARAMS, (int)f.number);
          return emit_varint(c, (uint64_t)v);
        }
        case K_SINT64: {
          int64_t v;
          if (!num_to_i64(nv, &v)) return fail(c, E_INVALID_PARAMS, (int)f.number);
          return emit_varint(c, zigzag64(v));
        }
        case K_SFIXED64: {
          int64_t v;
          if (!num_to_i64(nv, &v)) return fail(c, E_INVALID_PARAMS, (int)f.number);
          return emit_fixed64(c, (uint64_t)v);
        }
        case K_UINT64: {
          uint64_t v;
          if (!num_to_u64(nv, &v)) return fail(c, E_INVALID_PARAMS, (int)f.number);
          return emit_varint(c, v);
        }
        case K_FIXED64: {
          uint64_t v;
          if (!num_to_u64(nv, &v)) return fail(c, E_INVALID_PARAMS, (int)f.number);
          return emit_fixed64(c, v);
        }
        case K_INT32: {
          int64_t v;
          if (!num_to_i64(nv, &v) || v < -2147483648ll || v > 2147483647ll)
            return fail(c, E_INVALID_PARAMS, (int)f.number);
          return emit_varint(c, (uint64_t)v);
        }
        case K_SINT32: {
          int64_t v;
          if (!num_to_i64(nv, &v) || v < -2147483648ll || v > 2147483647ll)
            return fail(c, E_INVALID_PARAMS, (int)f.number);
          return emit_varint(c, zigzag32((int32_t)v));
        }
        case K_SFIXED32: {
          int64_t v;
          if (!num_to_i64(nv, &v) || v < -2147483648ll || v > 2147483647ll)
            return fail(c, E_INVALID_PARAMS, (int)f.number);
          return emit_fixed32(c, (uint32_t)(int32_t)v);
        }
        case K_UINT32: {
          uint64_t v;
          if (!num_to_u64(nv, &v) || v > 0xFFFFFFFFull)
            return fail(c, E_INVALID_PARAMS, (int)f.number);
          return emit_varint(c, v);
        }
        case K_FIXED32: {
          uint64_t v;
          if (!num_to_u64(nv, &v) || v > 0xFFFFFFFFull)
            return fail(c, E_INVALID_PARAMS, (int)f.number);
          return emit_fixed32(c, (uint32_t)v);
        }
      }
      return fail(c, E_UNSUPPORTED);
    }
  }
}

DEV bool is_packable(uint8_t kind) {
  return kind != K_STRING && kind != K_BYTES && kind != K_MESSAGE &&
         kind != K_GROUP;
}


// ---------------------------------------------------------------------------
// encoder — ITERATIVE walker (see pb2json.hip for the rationale: the
// recursive form kept the whole context in scratch memory, costing ~170
// cycles per processed byte; this flat walker keeps hot state in VGPRs and
// removes the dynamic device stack).  Nested-message length prefixes are
// 3-byte non-minimal varints backfilled on frame pop, so no sizing pass.
// ---------------------------------------------------------------------------

enum : uint8_t { EM_BODY = 0, EM_STRUCT = 1, EM_LIST = 2 };
// resume states
enum : uint8_t {
  RB_KEY = 0,     // EM_BODY: at a member key
  RB_SEP = 1,     // EM_BODY: after a member value (',' or '}')
  RB_ARR = 2,     // EM_BODY: at a repeated element
  RB_ARRSEP = 3,  // EM_BODY: after a repeated element
  RB_MAPKEY = 4,  // EM_BODY: at a map entry key
  RB_MAPSEP = 5,  // EM_BODY: after a map entry
  RS_KEY = 6,     // EM_STRUCT: at an entry key
  RS_SEP = 7,     // EM_STRUCT: after an entry
  RL_ELEM = 8,    // EM_LIST: at an element
  RL_SEP = 9,     // EM_LIST: after an element
};

struct EFrame {
  uint64_t seen_fields;
  uint32_t seen_oneofs;
  uint32_t slots[3];   // len slots backfilled (reverse order) on pop
  int32_t msg_idx;
  int32_t cont_field;  // absolute field index of the active array/map
  uint8_t n_slots;
  uint8_t mode;
  uint8_t resume;
};

// Value scalar one-shot: null/bool/string/number emit a complete Value
// payload (returns 1); '{' -> emits struct_value tag+slot, consumes '{',
// returns 2 (*slot set); '[' likewise returns 3.  0 = error.
DEVN int encode_value_scalar_or_classify(Ctx& c, uint32_t* slot) {
  skip_ws(c);
  uint8_t ch = peek(c);
  if (literal_at(c, "null", 4)) {
    c.pos += 4;
    if (!emit_tag(c, 1, W_VARINT)) return 0;
    return emit_varint(c, 0) ? 1 : 0;
  }
  if (literal_at(c, "true", 4)) {
    c.pos += 4;
    if (!emit_tag(c, 4, W_VARINT)) return 0;
    return emit_varint(c, 1) ? 1 : 0;
  }
  if (literal_at(c, "false", 5)) {
    c.pos += 5;
    if (!emit_tag(c, 4, W_VARINT)) return 0;
    return emit_varint(c, 0) ? 1 : 0;
  }
  if (ch == '"') {
    uint32_t st, rl;
    bool esc;
    if (!string_span(c, &st, &rl, &esc)) return 0;
    if (!emit_tag(c, 3, W_LEN)) return 0;
    return encode_string_payload(c, st, rl, esc) ? 1 : 0;
  }
  if (ch == '{') {
    if (!emit_tag(c, 5, W_LEN)) return 0;
    if (!reserve_len(c, slot)) return 0;
    c.pos++;
    return 2;
  }
  if (ch == '[') {
    if (!emit_tag(c, 6, W_LEN)) return 0;
    if (!reserve_len(c, slot)) return 0;
    c.pos++;
    return 3;
  }
  NumVal nv;
  int nonf;
  if (!parse_numeric_value(c, &nv, false, &nonf)) return 0;
  double d = nv.cls == 0 ? (nv.neg ? -(double)nv.mag : (double)nv.mag) : nv.d;
  // boundary decimals (DBL_MAX edge, >20 digits, near-denormal) -> host
  if (nv.imprecise) return fail(c, E_UNSUPPORTED) ? 1 : 0;
  if (!emit_tag(c, 2, W_I64)) return 0;
  return emit_fixed64(c, __builtin_bit_cast(uint64_t, d)) ? 1 : 0;
}

// FieldMask field: JSON "a.b,camelCase" -> repeated snake_case strings
// (message payload only; caller owns tag/len)
DEVN bool encode_fieldmask_payload(Ctx& c) {
  uint32_t st, rl;
  bool esc;
  if (!string_span(c, &st, &rl, &esc)) return false;
  if (esc) return fail(c, E_UNSUPPORTED);
  uint32_t i = 0;
  while (i < rl) {
    uint32_t j = i;
    while (j < rl && c.s[st + j] != ',') ++j;
    if (j > i) {
      if (!emit_tag(c, 1, W_LEN)) return false;
      uint32_t slot;
      if (!reserve_len(c, &slot)) return false;
      for (uint32_t k = i; k < j; ++k) {
        uint8_t ch = c.s[st + k];
        if (ch >= 'A' && ch <= 'Z') {
          if (c.opos + 2 > c.ocap) return fail(c, E_OVERFLOW);
          if (!c.lane) {
            c.out[c.opos] = '_';
            c.out[c.opos + 1] = ch + 32;
          }
          c.opos += 2;
        } else {
          if (c.opos + 1 > c.ocap) return fail(c, E_OVERFLOW);
          if (!c.lane) c.out[c.opos] = ch;
          c.opos += 1;
        }
      }
      if (!backfill_len(c, slot)) return false;
    }
    i = j + 1;
  }
  return true;
}

// Timestamp/Duration field payload from a JSON string (no tag/len)
DEVN bool encode_tsdur_payload(Ctx& c, bool is_ts, uint32_t fnum) {
  uint32_t st, rl;
  bool esc;
  if (!string_span(c, &st, &rl, &esc)) return false;
  if (esc) return fail(c, E_UNSUPPORTED);
  int64_t secs;
  int32_t nanos;
  bool ok = is_ts ? parse_timestamp(c, c.s + st, rl, &secs, &nanos)
                  : parse_duration(c.s + st, rl, &secs, &nanos);
  if (!ok) return fail(c, E_INVALID_PARAMS, (int)fnum);
  if (secs) {
    if (!emit_varint(c, (1u << 3) | W_VARINT)) return false;
    if (!emit_varint(c, (uint64_t)secs)) return false;
  }
  if (nanos) {
    if (!emit_varint(c, (2u << 3) | W_VARINT)) return false;
    if (!emit_varint(c, (uint64_t)(int64_t)nanos)) return false;
  }
  return true;
}

// map entry key: parse the JSON key span, emit entry tag + len slot + key
// field (1).  *eslot receives the entry slot.
DEVN bool encode_map_entry_key(Ctx& c, const FieldEntry& f, const FieldEntry& kf,
                               uint32_t* eslot) {
  uint32_t kst, krl;
  bool kesc;
  if (!string_span(c, &kst, &krl, &kesc)) return false;
  if (kesc) return fail(c, E_UNSUPPORTED);
  if (!expect(c, ':')) return false;
  if (!emit_tag(c, f.number, W_LEN)) return false;
  if (!reserve_len(c, eslot)) return false;
  switch (kf.kind) {
    case K_STRING: {
      if (!emit_tag(c, 1, W_LEN)) return false;
      return encode_string_payload(c, kst, krl, false);
    }
    case K_BOOL: {
      uint64_t v;
      if (krl == 4 && c.s[kst] == 't') v = 1;
      else if (krl == 5 && c.s[kst] == 'f') v = 0;
      else return fail(c, E_INVALID_PARAMS, (int)f.number);
      if (!emit_tag(c, 1, W_VARINT)) return false;
      return emit_varint(c, v);
    }
    default: {  // integer keys
      NumVal nv;
      if (!parse_number_text(c.s, kst, kst + krl, &nv))
        return fail(c, E_INVALID_PARAMS, (int)f.number);
      int64_t sv;
      uint64_t uv;
      switch (kf.kind) {
        case K_INT64:
        case K_SFIXED64:
        case K_SINT64:
        case K_INT32:
        case K_SINT32:
        case K_SFIXED32:
          if (!num_to_i64(nv, &sv)) return fail(c, E_INVALID_PARAMS, (int)f.number);
          break;
        default:
          if (!num_to_u64(nv, &uv)) return fail(c, E_INVALID_PARAMS, (int)f.number);
          sv = (int64_t)uv;
      }
      if (kf.kind == K_SINT64) uv = zigzag64(sv);
      else if (kf.kind == K_SINT32) uv = zigzag32((int32_t)sv);
      else uv = (uint64_t)sv;
      if (kf.kind == K_FIXED64 || kf.kind == K_SFIXED64) {
        if (!emit_tag(c, 1, W_I64)) return false;
        return emit_fixed64(c, uv);
      }
      if (kf.kind == K_FIXED32 || kf.kind == K_SFIXED32) {
        if (!emit_tag(c, 1, W_I32)) return false;
        return emit_fixed32(c, (uint32_t)uv);
      }
      if (!emit_tag(c, 1, W_VARINT)) return false;
      return emit_varint(c, uv);
    }
  }
}

// message-typed field value (tag NOT yet emitted).  Leaf WKTs complete
// inline (returns 1).  Container bodies emit tag + slot(s), consume the
// opening brace/bracket and return 2 with push parameters.  0 = error.
DEVN int encode_msgfield_value(Ctx& c, const FieldEntry& f, uint8_t* push_mode,
                               int32_t* push_idx, uint32_t* slots,
                               uint8_t* n_slots) {
  const MsgEntry& sub = c.t.msgs[f.sub_index];
  *n_slots = 0;
  switch (sub.wkt_kind) {
    case WKT_TIMESTAMP:
    case WKT_DURATION: {
      if (!emit_tag(c, f.number, W_LEN)) return 0;
      uint32_t slot;
      if (!reserve_len(c, &slot)) return 0;
      if (!encode_tsdur_payload(c, sub.wkt_kind == WKT_TIMESTAMP, f.number))
        return 0;
      return backfill_len(c, slot) ? 1 : 0;
    }
    case WKT_WRAPPER: {
      if (!emit_tag(c, f.number, W_LEN)) return 0;
      uint32_t slot;
      if (!reserve_len(c, &slot)) return 0;
      const FieldEntry& inner = c.t.fields[sub.field_start];
      if (!encode_scalar_field(c, inner)) return 0;
      return backfill_len(c, slot) ? 1 : 0;
    }
    case WKT_FIELDMASK: {
      if (!emit_tag(c, f.number, W_LEN)) return 0;
      uint32_t slot;
      if (!reserve_len(c, &slot)) return 0;
      if (!encode_fieldmask_payload(c)) return 0;
      return backfill_len(c, slot) ? 1 : 0;
    }
    case WKT_EMPTY: {
      if (!emit_tag(c, f.number, W_LEN)) return 0;
      uint32_t slot;
      if (!reserve_len(c, &slot)) return 0;
      if (!expect(c, '{')) return 0;
      skip_ws(c);
      if (!expect(c, '}')) return 0;
      return backfill_len(c, slot) ? 1 : 0;
    }
    case WKT_VALUE: {
      if (!emit_tag(c, f.number, W_LEN)) return 0;
      uint32_t vslot;
      if (!reserve_len(c, &vslot)) return 0;
      uint32_t inner_slot = 0;
      int r = encode_value_scalar_or_classify(c, &inner_slot);
      if (r == 0) return 0;
      if (r == 1) return backfill_len(c, vslot) ? 1 : 0;
      *push_mode = (r == 2) ? EM_STRUCT : EM_LIST;
      *push_idx = -1;
      slots[0] = inner_slot;
      slots[1] = vslot;
      *n_slots = 2;
      return 2;
    }
    case WKT_STRUCT: {
      if (!emit_tag(c, f.number, W_LEN)) return 0;
      uint32_t slot;
      if (!reserve_len(c, &slot)) return 0;
      if (!expect(c, '{')) return 0;
      *push_mode = EM_STRUCT;
      *push_idx = -1;
      slots[0] = slot;
      *n_slots = 1;
      return 2;
    }
    case WKT_LISTVALUE: {
      if (!emit_tag(c, f.number, W_LEN)) return 0;
      uint32_t slot;
      if (!reserve_len(c, &slot)) return 0;
      if (!expect(c, '[')) return 0;
      *push_mode = EM_LIST;
      *push_idx = -1;
      slots[0] = slot;
      *n_slots = 1;
      return 2;
    }
    case WKT_ANY:
      return fail(c, E_UNSUPPORTED) ? 1 : 0;
    default: {  // plain nested message
      if (!emit_tag(c, f.number, W_LEN)) return 0;
      uint32_t slot;
      if (!reserve_len(c, &slot)) return 0;
      if (!expect(c, '{')) return 0;
      *push_mode = EM_BODY;
      *push_idx = f.sub_index;
      slots[0] = slot;
      *n_slots = 1;
      return 2;
    }
  }
}

// member key lookup inside EM_BODY (hash + dup/oneof bookkeeping);
// returns field index within the message or -1 (error already set)
DEVN int encode_body_key(Ctx& c, EFrame& f) {
  const MsgEntry& m = c.t.msgs[f.msg_idx];
  uint32_t kst, krl;
  bool kesc;
  if (!string_span(c, &kst, &krl, &kesc)) return -1;
  const uint8_t* kptr = c.s + kst;
  uint32_t klen = krl;
  if (kesc) {
    uint32_t n = unescape_serial(c, c.s + kst, krl, c.keybuf, 192);
    if (n == 0xFFFFFFFF) {
      fail(c, E_PARSE);
      return -1;
    }
    __builtin_amdgcn_wave_barrier();
    kptr = c.keybuf;
    klen = n;
  }
  if (!expect(c, ':')) return -1;
  uint64_t h = fnv1a64(kptr, klen);
  int fidx = -1;
  for (int i = 0; i < m.field_count; ++i) {
    const FieldEntry& fe = c.t.fields[m.field_start + i];
    if ((fe.hash_json == h && fe.json_len == klen &&
         wave_equal(c, c.t.names + fe.json_off, kptr, klen)) ||
        (fe.hash_orig == h && fe.name_len == klen &&
         wave_equal(c, c.t.names + fe.name_off, kptr, klen))) {
      fidx = i;
      break;
    }
  }
  if (fidx < 0) {
    c.err_pos = kst;
    fail(c, E_INVALID_PARAMS, -1);  // unknown field: protojson rejects
    return -1;
  }
  const FieldEntry& fe = c.t.fields[m.field_start + fidx];
  if (fidx < 64) {
    if (f.seen_fields & (1ull << fidx)) {
      fail(c, E_INVALID_PARAMS, (int)fe.number);
      return -1;
    }
    f.seen_fields |= 1ull << fidx;
  }
  if (fe.flags & F_ONEOF) {
    if (f.seen_oneofs & (1u << fe.oneof_id)) {
      fail(c, E_INVALID_PARAMS, (int)fe.number);
      return -1;
    }
    f.seen_oneofs |= 1u << fe.oneof_id;
  }
  return m.field_start + fidx;
}

// whole packed repeated array "[e,e,...]" — '[' already consumed, at first
// element; emits tag + len slot + packed payload + backfill
DEVN bool encode_packed_array(Ctx& c, const FieldEntry& f) {
  if (!emit_tag(c, f.number, W_LEN)) return false;
  uint32_t slot;
  if (!reserve_len(c, &slot)) return false;
  while (true) {
    skip_ws(c);
    if (!encode_packed_element(c, f)) return false;
    skip_ws(c);
    uint8_t ch = peek(c);
    if (ch == ',') {
      c.pos++;
      continue;
    }
    if (ch == ']') {
      c.pos++;
      break;
    }
    return fail(c, E_PARSE);
  }
  return backfill_len(c, slot);
}

// wg_item: workgroup-encode item modes (k_json2pb_wg). 0 = classic whole
// value; 1 = one top-level member `"key": value`; 2 = map-entry span of
// field wg_fidx (`"k":v, ...`, no braces); 3 = array-element span of
// wg_fidx.  Item spans END the walk at c.len instead of a closing brace.
DEV bool encode_walk(Ctx& c, int top_msg_idx, int wg_item = 0,
                     int wg_fidx = -1) {
  EFrame stack[MAX_RECURSE];
  int sp = 0;
  int depth_cap = (int)MAX_RECURSE;
  if (c.lim.enforce && (int)c.lim.max_depth < depth_cap)
    depth_cap = (int)c.lim.max_depth;

#define EPUSH(MODE, MIDX, SLOTS_ARR, NS)                      \
  do {                                                        \
    if (sp >= depth_cap) return fail(c, E_LIMIT, 1);          \
    EFrame& nf = stack[sp++];                                 \
    nf.seen_fields = 0;                                       \
    nf.seen_oneofs = 0;                                       \
    nf.msg_idx = (MIDX);                                      \
    nf.cont_field = -1;                                       \
    nf.n_slots = (NS);                                        \
    for (int _i = 0; _i < (NS); ++_i) nf.slots[_i] = (SLOTS_ARR)[_i]; \
    nf.mode = (MODE);                                         \
    nf.resume = (MODE) == EM_BODY ? RB_KEY                    \
                : (MODE) == EM_STRUCT ? RS_KEY : RL_ELEM;     \
  } while (0)

  // ---- top-level entry ----
  if (wg_item) {
    uint32_t slots0[3];
    EPUSH(EM_BODY, top_msg_idx, slots0, 0);
    if (wg_item == 2) {
      stack[0].cont_field = wg_fidx;
      stack[0].resume = RB_MAPKEY;
    } else if (wg_item == 3) {
      stack[0].cont_field = wg_fidx;
      stack[0].resume = RB_ARR;
    }
    // RB_KEY expects the cursor at the member's key quote
  } else {
    const MsgEntry& m = c.t.msgs[top_msg_idx];
    uint32_t slots0[3];
    if (m.wkt_kind == WKT_VALUE) {
      uint32_t islot = 0;
      int r = encode_value_scalar_or_classify(c, &islot);
      if (r == 0) return false;
      if (r == 1) return true;
      slots0[0] = islot;
      EPUSH(r == 2 ? EM_STRUCT : EM_LIST, -1, slots0, 1);
      // fall into loop; empty-body handled by the frame steps
      skip_ws(c);
      if (r == 2 && peek(c) == '}') {
        c.pos++;
        if (!backfill_len(c, islot)) return false;
        --sp;
      } else if (r == 3 && peek(c) == ']') {
        c.pos++;
        if (!backfill_len(c, islot)) return false;
        --sp;
      }
    } else if (m.wkt_kind == WKT_STRUCT) {
      if (!expect(c, '{')) return false;
      skip_ws(c);
      if (peek(c) == '}') {
        c.pos++;
        return true;
      }
      EPUSH(EM_STRUCT, -1, slots0, 0);
    } else if (m.wkt_kind == WKT_LISTVALUE) {
      if (!expect(c, '[')) return false;
      skip_ws(c);
      if (peek(c) == ']') {
        c.pos++;
        return true;
      }
      EPUSH(EM_LIST, -1, slots0, 0);
    } else if (m.wkt_kind == WKT_FIELDMASK) {
      return encode_fieldmask_payload(c);
    } else if (m.wkt_kind == WKT_EMPTY) {
      if (!expect(c, '{')) return false;
      skip_ws(c);
      return expect(c, '}');
    } else {
      if (!expect(c, '{')) return false;
      skip_ws(c);
      if (peek(c) == '}') {
        c.pos++;
        return true;
      }
      EPUSH(EM_BODY, top_msg_idx, slots0, 0);
    }
  }

  while (sp > 0) {
    EFrame& f = stack[sp - 1];
    switch (f.resume) {
      case RB_KEY: {
        int fidx = encode_body_key(c, f);
        if (fidx < 0) return false;
        const FieldEntry& fe = c.t.fields[fidx];
        skip_ws(c);
        bool is_value_wkt =
            fe.kind == K_MESSAGE && c.t.msgs[fe.sub_index].wkt_kind == WKT_VALUE;
        if (literal_at(c, "null", 4) && !is_value_wkt) {
          c.pos += 4;  // null -> unset (protojson)
          f.resume = RB_SEP;
          continue;
        }
        if (fe.flags & F_MAP) {
          if (!expect(c, '{')) return false;
          skip_ws(c);
          if (peek(c) == '}') {
            c.pos++;
            f.resume = RB_SEP;
            continue;
          }
          f.cont_field = fidx;
          f.resume = RB_MAPKEY;
          continue;
        }
        if (fe.flags & F_REPEATED) {
          if (!expect(c, '[')) return false;
          skip_ws(c);
          if (peek(c) == ']') {
            c.pos++;
            f.resume = RB_SEP;
            continue;
          }
          if (is_packable(fe.kind)) {
            if (!encode_packed_array(c, fe)) return false;
            f.resume = RB_SEP;
            continue;
          }
          f.cont_field = fidx;
          f.resume = RB_ARR;
          continue;
        }
        // singular
        if (fe.kind == K_MESSAGE) {
          uint8_t pm;
          int32_t pidx;
          uint32_t slots[3];
          uint8_t ns;
          int r = encode_msgfield_value(c, fe, &pm, &pidx, slots, &ns);
          if (r == 0) return false;
          if (r == 1) {
            f.resume = RB_SEP;
            continue;
          }
          f.resume = RB_SEP;
          EPUSH(pm, pidx, slots, ns);
          // empty body fast path
          skip_ws(c);
          if ((pm != EM_LIST && peek(c) == '}') ||
              (pm == EM_LIST && peek(c) == ']')) {
            c.pos++;
            EFrame& nf = stack[sp - 1];
            for (int i = 0; i < nf.n_slots; ++i)
              if (!backfill_len(c, nf.slots[i])) return false;
            --sp;
          }
          continue;
        }
        if (!encode_scalar_field(c, fe)) return false;
        f.resume = RB_SEP;
        continue;
      }
      case RB_SEP: {
        skip_ws(c);
        if (wg_item == 1 && sp == 1 && c.pos >= c.len) {
          --sp;  // member item ends at its span, no brace to consume
          continue;
        }
        uint8_t ch = peek(c);
        if (ch == ',') {
          c.pos++;
          skip_ws(c);
          f.resume = RB_KEY;
          continue;
        }
        if (ch == '}') {
          c.pos++;
          for (int i = 0; i < f.n_slots; ++i)
            if (!backfill_len(c, f.slots[i])) return false;
          --sp;
          continue;
        }
        return fail(c, E_PARSE);
      }
      case RB_ARR: {
        const FieldEntry& fe = c.t.fields[f.cont_field];
        skip_ws(c);
        if (fe.kind == K_MESSAGE) {
          uint8_t pm;
          int32_t pidx;
          uint32_t slots[3];
          uint8_t ns;
          int r = encode_msgfield_value(c, fe, &pm, &pidx, slots, &ns);
          if (r == 0) return false;
          if (r == 1) {
            f.resume = RB_ARRSEP;
            continue;
          }
          f.resume = RB_ARRSEP;
          EPUSH(pm, pidx, slots, ns);
          skip_ws(c);
          if ((pm != EM_LIST && peek(c) == '}') ||
              (pm == EM_LIST && peek(c) == ']')) {
            c.pos++;
            EFrame& nf = stack[sp - 1];
            for (int i = 0; i < nf.n_slots; ++i)
              if (!backfill_len(c, nf.slots[i])) return false;
            --sp;
          }
          continue;
        }
        if (!encode_scalar_field(c, fe)) return false;
        f.resume = RB_ARRSEP;
        continue;
      }
      case RB_ARRSEP: {
        skip_ws(c);
        if (wg_item == 3 && sp == 1 && c.pos >= c.len) {
          --sp;  // element-span item: no ']' in the span
          continue;
        }
        uint8_t ch = peek(c);
        if (ch == ',') {
          c.pos++;
          f.resume = RB_ARR;
          continue;
        }
        if (ch == ']') {
          c.pos++;
          f.resume = RB_SEP;
          continue;
        }
        return fail(c, E_PARSE);
      }
      case RB_MAPKEY: {
        const FieldEntry& fe = c.t.fields[f.cont_field];
        const MsgEntry& em = c.t.msgs[fe.sub_index];
        const FieldEntry& kf = c.t.fields[em.field_start];
        const FieldEntry& vf = c.t.fields[em.field_start + 1];
        uint32_t eslot;
        if (!encode_map_entry_key(c, fe, kf, &eslot)) return false;
        skip_ws(c);
        if (literal_at(c, "null", 4) && vf.kind != K_MESSAGE) {
          c.pos += 4;  // null map value -> default (key-only entry)
          if (!backfill_len(c, eslot)) return false;
          f.resume = RB_MAPSEP;
          continue;
        }
        if (vf.kind == K_MESSAGE) {
          // value tag is field 2 inside the entry
          FieldEntry vf2 = vf;  // value field with number 2 by construction
          uint8_t pm;
          int32_t pidx;
          uint32_t slots[3];
          uint8_t ns;
          int r = encode_msgfield_value(c, vf2, &pm, &pidx, slots, &ns);
          if (r == 0) return false;
          if (r == 1) {
            if (!backfill_len(c, eslot)) return false;
            f.resume = RB_MAPSEP;
            continue;
          }
          if (ns >= 3) return fail(c, E_LIMIT, 1);
          slots[ns++] = eslot;  // entry slot backfilled last
          f.resume = RB_MAPSEP;
          EPUSH(pm, pidx, slots, ns);
          skip_ws(c);
          if ((pm != EM_LIST && peek(c) == '}') ||
              (pm == EM_LIST && peek(c) == ']')) {
            c.pos++;
            EFrame& nf = stack[sp - 1];
            for (int i = 0; i < nf.n_slots; ++i)
              if (!backfill_len(c, nf.slots[i])) return false;
            --sp;
          }
          continue;
        }
        if (!encode_scalar_field(c, vf)) return false;
        if (!backfill_len(c, eslot)) return false;
        f.resume = RB_MAPSEP;
        continue;
      }
      case RB_MAPSEP: {
        skip_ws(c);
        if (wg_item == 2 && sp == 1 && c.pos >= c.len) {
          --sp;  // entry-span item: no '}' in the span
          continue;
        }
        uint8_t ch = peek(c);
        if (ch == ',') {
          c.pos++;
          skip_ws(c);
          f.resume = RB_MAPKEY;
          continue;
        }
        if (ch == '}') {
          c.pos++;
          f.resume = RB_SEP;
          continue;
        }
        return fail(c, E_PARSE);
      }
      case RS_KEY: {
        uint32_t kst, krl;
        bool kesc;
        if (!string_span(c, &kst, &krl, &kesc)) return false;
        if (!expect(c, ':')) return false;
        if (!emit_tag(c, 1, W_LEN)) return false;
        uint32_t eslot;
        if (!reserve_len(c, &eslot)) return false;
        if (!emit_tag(c, 1, W_LEN)) return false;
        if (!encode_string_payload(c, kst, krl, kesc)) return false;
        if (!emit_tag(c, 2, W_LEN)) return false;
        uint32_t vslot;
        if (!reserve_len(c, &vslot)) return false;
        uint32_t islot = 0;
        int r = encode_value_scalar_or_classify(c, &islot);
        if (r == 0) return false;
        if (r == 1) {
          if (!backfill_len(c, vslot)) return false;
          if (!backfill_len(c, eslot)) return false;
          f.resume = RS_SEP;
          continue;
        }
        uint32_t slots[3] = {islot, vslot, eslot};
        f.resume = RS_SEP;
        EPUSH(r == 2 ? EM_STRUCT : EM_LIST, -1, slots, 3);
        skip_ws(c);
        if ((r == 2 && peek(c) == '}') || (r == 3 && peek(c) == ']')) {
          c.pos++;
          EFrame& nf = stack[sp - 1];
          for (int i = 0; i < nf.n_slots; ++i)
            if (!backfill_len(c, nf.slots[i])) return false;
          --sp;
        }
        continue;
      }
      case RS_SEP: {
        skip_ws(c);
        uint8_t ch = peek(c);
        if (ch == ',') {
          c.pos++;
          skip_ws(c);
          f.resume = RS_KEY;
          continue;
        }
        if (ch == '}') {
          c.pos++;
          for (int i = 0; i < f.n_slots; ++i)
            if (!backfill_len(c, f.slots[i])) return false;
          --sp;
          continue;
        }
        return fail(c, E_PARSE);
      }
      case RL_ELEM: {
        if (!emit_tag(c, 1, W_LEN)) return false;
        uint32_t vslot;
        if (!reserve_len(c, &vslot)) return false;
        uint32_t islot = 0;
        int r = encode_value_scalar_or_classify(c, &islot);
        if (r == 0) return false;
        if (r == 1) {
          if (!backfill_len(c, vslot)) return false;
          f.resume = RL_SEP;
          continue;
        }
        uint32_t slots[2] = {islot, vslot};
        f.resume = RL_SEP;
        EPUSH(r == 2 ? EM_STRUCT : EM_LIST, -1, slots, 2);
        skip_ws(c);
        if ((r == 2 && peek(c) == '}') || (r == 3 && peek(c) == ']')) {
          c.pos++;
          EFrame& nf = stack[sp - 1];
          for (int i = 0; i < nf.n_slots; ++i)
            if (!backfill_len(c, nf.slots[i])) return false;
          --sp;
        }
        continue;
      }
      case RL_SEP: {
        skip_ws(c);
        uint8_t ch = peek(c);
        if (ch == ',') {
          c.pos++;
          f.resume = RL_ELEM;
          continue;
        }
        if (ch == ']') {
          c.pos++;
          for (int i = 0; i < f.n_slots; ++i)
            if (!backfill_len(c, f.slots[i])) return false;
          --sp;
          continue;
        }
        return fail(c, E_PARSE);
      }
    }
  }
#undef EPUSH
  return c.status == E_OK;
}

// ---------------------------------------------------------------------------
// envelope (mode 0): full JSON-RPC tools/call request
// ---------------------------------------------------------------------------

// wg_args non-null: stop after envelope validation + tool resolution and
// return the arguments span as {pos, end} (0xFFFFFFFF = absent) instead of
// encoding — the workgroup encode kernel splits the arguments itself.
DEV bool parse_envelope(Ctx& c, SlotResult& r, uint8_t* id_slot,
                        uint32_t* wg_args = nullptr) {
  bool saw_jsonrpc = false, jsonrpc_ok = false;
  bool saw_method = false, method_ok = false;
  bool saw_id = false;
  bool saw_params = false;
  uint32_t name_start = 0, name_len = 0;
  bool have_name = false;
  uint32_t args_pos = 0xFFFFFFFF, args_end = 0;

  if (!expect(c, '{')) return fail(c, E_PARSE);
  skip_ws(c);
  if (peek(c) == '}') return fail(c, E_INVALID_REQUEST, 1);
  while (true) {
    uint32_t kst, krl;
    bool kesc;
    if (!string_span(c, &kst, &krl, &kesc)) return false;
    if (!expect(c, ':')) return false;
    skip_ws(c);
    const uint8_t* k = c.s + kst;
    if (!kesc && krl == 7 && k[0] == 'j' && wave_equal(c, k, (const uint8_t*)"jsonrpc", 7)) {
      saw_jsonrpc = true;
      uint32_t vst, vrl;
      bool vesc;
      if (!string_span(c, &vst, &vrl, &vesc)) return false;
      jsonrpc_ok = !vesc && vrl == 3 && c.s[vst] == '2' && c.s[vst + 1] == '.' &&
                   c.s[vst + 2] == '0';
    } else if (!kesc && krl == 2 && k[0] == 'i' && k[1] == 'd') {
      saw_id = true;
      uint32_t v0 = c.pos;
      if (!skip_value(c)) return false;
      uint32_t vlen = c.pos - v0;
      if (vlen > ID_SLOT_BYTES) return fail(c, E_UNSUPPORTED, 2);
      wave_copy(c, id_slot, c.s + v0, vlen);
      r.id_len = vlen;
    } else if (!kesc && krl == 6 && k[0] == 'm' &&
               wave_equal(c, k, (const uint8_t*)"method", 6)) {
      saw_method = true;
      uint32_t vst, vrl;
      bool vesc;
      if (!string_span(c, &vst, &vrl, &vesc)) return false;
      method_ok = !vesc && vrl == 10 &&
                  wave_equal(c, c.s + vst, (const uint8_t*)"tools/call", 10);
    } else if (!kesc && krl == 6 && k[0] == 'p' &&
               wave_equal(c, k, (const uint8_t*)"params", 6)) {
      saw_params = true;
      if (!expect(c, '{')) return fail(c, E_INVALID_PARAMS);
      skip_ws(c);
      if (peek(c) == '}') {
        c.pos++;
      } else {
        while (true) {
          uint32_t pkst, pkrl;
          bool pkesc;
          if (!string_span(c, &pkst, &pkrl, &pkesc)) return false;
          if (!expect(c, ':')) return false;
          skip_ws(c);
          const uint8_t* pk = c.s + pkst;
          if (!pkesc && pkrl == 4 && wave_equal(c, pk, (const uint8_t*)"name", 4)) {
            uint32_t nst, nrl;
            bool nesc;
            if (!string_span(c, &nst, &nrl, &nesc)) return false;
            if (nesc || nrl == 0 || nrl > 128) return fail(c, E_INVALID_REQUEST, 3);
            name_start = nst;
            name_len = nrl;
            have_name = true;
          } else if (!pkesc && pkrl == 9 &&
                     wave_equal(c, pk, (const uint8_t*)"arguments", 9)) {
            args_pos = c.pos;
            if (!skip_value(c)) return false;
            args_end = c.pos;
          } else {
            if (!skip_value(c)) return false;  // _meta etc: tolerated
          }
          skip_ws(c);
          uint8_t ch = peek(c);
          if (ch == ',') {
            c.pos++;
            skip_ws(c);
            continue;
          }
          if (ch == '}') {
            c.pos++;
            break;
          }
          return fail(c, E_PARSE);
        }
      }
    } else {
      if (!skip_value(c)) return false;  // unknown envelope member: tolerated
    }
    skip_ws(c);
    uint8_t ch = peek(c);
    if (ch == ',') {
      c.pos++;
      skip_ws(c);
      continue;
    }
    if (ch == '}') {
      c.pos++;
      break;
    }
    return fail(c, E_PARSE);
  }
  // trailing garbage check
  skip_ws(c);
  if (c.pos < c.len) return fail(c, E_PARSE);

  // ---- envelope verdicts (validation.go:24-61, 96-125) ----
  if (!saw_jsonrpc || !jsonrpc_ok) return fail(c, E_INVALID_REQUEST, 4);
  if (!saw_method) return fail(c, E_INVALID_REQUEST, 5);
  if (!saw_id) r.flags |= SR_ID_IS_MISSING;
  if (!method_ok) return fail(c, E_NOT_TOOLCALL);
  if (!saw_params || !have_name) return fail(c, E_INVALID_REQUEST, 6);

  // ---- tool lookup (discovery.go:336-343) ----
  uint64_t h = fnv1a64(c.s + name_start, name_len);
  int tool = -1;
  for (int i = 0; i < c.t.n_tools; ++i) {
    const ToolEntry& te = c.t.tools[i];
    if (te.hash == h && te.name_len == name_len &&
        wave_equal(c, c.t.names + te.name_off, c.s + name_start, name_len)) {
      tool = i;
      break;
    }
  }
  if (tool < 0) return fail(c, E_METHOD_NOT_FOUND);
  r.tool_idx = tool;
  if (c.t.tools[tool].flags & 1) r.flags |= SR_SERVER_STREAMING;

  // ---- encode arguments ----
  if (args_pos != 0xFFFFFFFF && c.lim.enforce &&
      args_end - args_pos > c.lim.max_args_bytes)
    return fail(c, E_LIMIT, 3);
  if (wg_args) {
    wg_args[0] = args_pos;
    wg_args[1] = args_end;
    return true;
  }
  if (args_pos == 0xFFFFFFFF) return true;  // no arguments -> empty message
  uint32_t save_len = c.len;
  c.pos = args_pos;
  c.len = args_end;
  bool ok = encode_walk(c, c.t.tools[tool].in_msg);
  c.len = save_len;
  return ok;
}

// ---------------------------------------------------------------------------
// kernel
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(WPB * WAVE) k_json2pb(
    const uint8_t* __restrict__ in_bytes, const uint32_t* __restrict__ in_off,
    uint8_t* __restrict__ pb_arena, const uint32_t* __restrict__ pb_off,
    SlotResult* __restrict__ results, uint8_t* __restrict__ id_slots,
    const int32_t* __restrict__ msg_idx_in, Tables t, Limits lim, int n_req,
    int mode, const int32_t* __restrict__ enc_skip) {
  __shared__ uint8_t keybufs[WPB][192];
  int wave_in_block = threadIdx.x / WAVE;
  for (int req = blockIdx.x * WPB + wave_in_block; req < n_req;
       req += gridDim.x * WPB) {
    if (enc_skip && enc_skip[req]) continue;  // k_json2pb_wg owns this slot
    Ctx c;
    c.s = in_bytes + in_off[req];
    c.len = in_off[req + 1] - in_off[req];
    c.pos = 0;
    c.out = pb_arena + pb_off[req];
    c.opos = 0;
    c.ocap = pb_off[req + 1] - pb_off[req];
    c.t = t;
    c.lim = lim;
    c.status = E_OK;
    c.err_pos = 0;
    c.aux = 0;
    c.lane = lane_id();
    c.keybuf = keybufs[wave_in_block];

    SlotResult r;
    r.status = E_OK;
    r.tool_idx = -1;
    r.pb_off = pb_off[req];
    r.pb_len = 0;
    r.err_pos = 0;
    r.aux = 0;
    r.id_len = 0;
    r.flags = 0;

    bool ok;
    if (mode == 0) {
      ok = parse_envelope(c, r, id_slots + (size_t)req * ID_SLOT_BYTES);
    } else {
      int msg_idx = msg_idx_in ? msg_idx_in[req] : 0;
      const MsgEntry& m = t.msgs[msg_idx];
      ok = encode_walk(c, msg_idx);
      if (ok) {
        skip_ws(c);
        if (c.pos < c.len) ok = fail(c, E_PARSE);
      }
    }
    (void)ok;
    r.status = c.status;
    r.err_pos = c.err_pos;
    r.aux = c.aux;
    r.pb_len = c.status == E_OK ? c.opos : 0;
    if (!c.lane) results[req] = r;
  }
}

#ifndef GGRMCP_HOST_SIM
// ---------------------------------------------------------------------------
// k_json2pb_wg — workgroup-per-request ENCODE for large requests
// (BASELINE config 3: 64 KB JSON bodies).
//
// One workgroup (WG_ENC_WAVES wave64s) owns one request:
//   A. wave 0 parses/validates the JSON-RPC envelope and resolves the tool
//      (parse_envelope with wg_args), then scans the arguments object into
//      top-level-member items; map members and non-packed repeated members
//      chunk at entry/element boundaries (~4 KB) — protobuf wire
//      CONCATENATION makes per-chunk emission valid with no joining
//      fix-ups (repeated/map runs may split across tags).
//   B. waves grab items dynamically and encode each with the SAME
//      encode_walk machinery (wg_item modes) into private scratch regions
//      carved from the decode scratch arena (idle during encode).
//   C. one thread prefix-sums item lengths; D. waves compact the items
//      into the slot's pb arena span.
// Any anomaly (escaped member keys, duplicate fields, item errors, caps)
// falls back to the classic single-wave encode inside the block, which
// reproduces the exact classic error formats.  The engine routes mode-0
// slots >= WG_ENC_MIN_BYTES here (skip tag 2).
// ---------------------------------------------------------------------------

// classic whole-slot mode-0 encode (the in-block fallback path); mirrors
// the k_json2pb loop body exactly
DEV void wg_enc_classic_one(const uint8_t* src, uint32_t src_len,
                            uint8_t* pbout, uint32_t pbcap,
                            SlotResult* results, uint8_t* id_slot,
                            uint8_t* keybuf, Tables t, Limits lim, int req,
                            int lane) {
  Ctx c;
  c.s = src;
  c.len = src_len;
  c.pos = 0;
  c.out = pbout;
  c.opos = 0;
  c.ocap = pbcap;
  c.t = t;
  c.lim = lim;
  c.status = E_OK;
  c.err_pos = 0;
  c.aux = 0;
  c.lane = lane;
  c.keybuf = keybuf;
  SlotResult r;
  r.status = E_OK;
  r.tool_idx = -1;
  r.pb_off = 0;  // caller fixes the arena offset
  r.pb_len = 0;
  r.err_pos = 0;
  r.aux = 0;
  r.id_len = 0;
  r.flags = 0;
  (void)parse_envelope(c, r, id_slot);
  r.status = c.status;
  r.err_pos = c.err_pos;
  r.aux = c.aux;
  r.pb_len = c.status == E_OK ? c.opos : 0;
  if (!lane) {
    results[req].status = r.status;
    results[req].tool_idx = r.tool_idx;
    results[req].pb_len = r.pb_len;
    results[req].err_pos = r.err_pos;
    results[req].aux = r.aux;
    results[req].id_len = r.id_len;
    results[req].flags = r.flags;
  }
}

extern "C" __global__ void __launch_bounds__(WG_ENC_WAVES * WAVE) k_json2pb_wg(
    const uint8_t* __restrict__ in_bytes, const uint32_t* __restrict__ in_off,
    uint8_t* __restrict__ pb_arena, const uint32_t* __restrict__ pb_off,
    SlotResult* __restrict__ results, uint8_t* __restrict__ id_slots,
    uint8_t* __restrict__ enc_scratch, Tables t, Limits lim, int n_req,
    const int32_t* __restrict__ skip, int max_phase, int use_mw) {
  // max_phase: debug bisection (GGRMCP_WG_ENC_PHASES; 3 = full)
  int req = blockIdx.x;
  if (req >= n_req) return;
  if (!skip || skip[req] != 2) return;

  __shared__ uint32_t s_start[WG_ENC_MAX_ITEMS];
  __shared__ uint32_t s_end[WG_ENC_MAX_ITEMS];
  __shared__ uint32_t s_ioff[WG_ENC_MAX_ITEMS];
  __shared__ uint32_t s_olen[WG_ENC_MAX_ITEMS];
  __shared__ uint32_t s_foff[WG_ENC_MAX_ITEMS];
  __shared__ int32_t s_fidx[WG_ENC_MAX_ITEMS];
  __shared__ uint8_t s_kind[WG_ENC_MAX_ITEMS];
  __shared__ int s_nitems, s_next, s_mode, s_msg;
  __shared__ uint32_t s_args0, s_args1;
  __shared__ SlotResult s_res;
  __shared__ uint8_t keybufs[WG_ENC_WAVES][192];

  const uint8_t* src = in_bytes + in_off[req];
  const uint32_t src_len = in_off[req + 1] - in_off[req];
  uint8_t* pbout = pb_arena + pb_off[req];
  const uint32_t pbcap = pb_off[req + 1] - pb_off[req];
  // scratch staging: 2x the slot's pb span (engine verified capacity)
  uint8_t* scr = enc_scratch + 2ull * pb_off[req];
  const uint32_t scr_cap = 2u * pbcap;
  const int lane = lane_id();
  const int wave = threadIdx.x / WAVE;

  // ---- phase A: envelope + member scan (wave 0: the string helpers are
  // wave-cooperative ballots) ----------------------------------------------
  if (wave == 0) {
    Ctx c;
    c.s = src;
    c.len = src_len;
    c.pos = 0;
    c.out = pbout;
    c.opos = 0;
    c.ocap = pbcap;
    c.t = t;
    c.lim = lim;
    c.status = E_OK;
    c.err_pos = 0;
    c.aux = 0;
    c.lane = lane;
    c.keybuf = keybufs[0];
    SlotResult r;
    r.status = E_OK;
    r.tool_idx = -1;
    r.pb_off = 0;
    r.pb_len = 0;
    r.err_pos = 0;
    r.aux = 0;
    r.id_len = 0;
    r.flags = 0;
    uint32_t args[2] = {0xFFFFFFFFu, 0};
    bool ok = parse_envelope(c, r, id_slots + (size_t)req * ID_SLOT_BYTES,
                             args);
    int mode_l;
    int n = 0;
    int msg_idx = 0;
    if (!ok || c.status != E_OK) {
      r.status = c.status;
      r.err_pos = c.err_pos;
      r.aux = c.aux;
      mode_l = -1;  // final: error envelope (host/kernel error mapping)
    } else if (args[0] == 0xFFFFFFFFu) {
      mode_l = -1;  // no arguments -> empty message, final
    } else {
      msg_idx = t.tools[r.tool_idx].in_msg;
      const MsgEntry& m = t.msgs[msg_idx];
      c.pos = args[0];
      c.len = args[1];
      uint64_t seen = 0;
      uint32_t seen_oneof = 0;
      uint32_t acc = 0;
      bool fallback = m.wkt_kind != WKT_NONE;
      mode_l = 1;
      // phase A' eligibility: big arguments objects index in parallel
      // across all waves (below); small ones keep the serial walk here
      if (!fallback && use_mw && args[1] - args[0] >= 8192) {
        mode_l = 2;
      } else if (!fallback && expect(c, '{')) {
        skip_ws(c);
        if (peek(c) == '}') {
          c.pos++;  // empty arguments object: zero items
        } else {
          while (true) {
            skip_ws(c);
            uint32_t kst, krl;
            bool kesc;
            if (!string_span(c, &kst, &krl, &kesc) || kesc) {
              fallback = true;  // escaped keys: classic handles
              break;
            }
            uint64_t h = fnv1a64(c.s + kst, krl);
            int fidx = -1;
            for (int i = 0; i < m.field_count; ++i) {
              const FieldEntry& fe = t.fields[m.field_start + i];
              if ((fe.hash_json == h && fe.json_len == krl &&
                   wave_equal(c, t.names + fe.json_off, c.s + kst, krl)) ||
                  (fe.hash_orig == h && fe.name_len == krl &&
                   wave_equal(c, t.names + fe.name_off, c.s + kst, krl))) {
                fidx = i;
                break;
              }
            }
            if (fidx < 0) {  // unknown field: classic formats the error
              fallback = true;
              break;
            }
            const FieldEntry& fe = t.fields[m.field_start + fidx];
            if (fidx < 64) {
              if (seen & (1ull << fidx)) {
                fallback = true;  // duplicate member -> classic error
                break;
              }
              seen |= 1ull << fidx;
            }
            if (fe.flags & F_ONEOF) {
              if (seen_oneof & (1u << fe.oneof_id)) {
                fallback = true;
                break;
              }
              seen_oneof |= 1u << fe.oneof_id;
            }
            if (!expect(c, ':')) {
              fallback = true;
              break;
            }
            skip_ws(c);
            uint32_t member_start = kst - 1;  // include the opening quote
            bool chunk_map = (fe.flags & F_MAP) && peek(c) == '{';
            bool chunk_arr = (fe.flags & F_REPEATED) &&
                             !(fe.flags & F_MAP) && !is_packable(fe.kind) &&
                             peek(c) == '[';
            bool big = (c.len - c.pos) >= 2048;
            if ((chunk_map || chunk_arr) && big) {
              // per-entry chunks: ONE bulk structural scan finds the
              // ~4 KB entry boundaries and the closer (the per-entry
              // string_span/skip_value loop paid a dependent memory
              // round trip per token); spans cover entries only — chunks
              // emit separate-but-concatenable runs
              c.pos++;  // consume '{' / '['
              skip_ws(c);
              uint8_t closer = chunk_map ? '}' : ']';
              if (peek(c) == closer) {
                c.pos++;  // empty container member: no wire output at all
              } else {
                uint32_t chunk_start = c.pos;
                uint32_t marks[64];
                int nm = 0;
                if (!scan_container(c, marks, &nm, 64, 4096)) {
                  fallback = true;
                  break;
                }
                uint32_t close_pos = c.pos - 1;
                if (c.s[close_pos] != closer) {  // '}' vs ']' mismatch
                  fallback = true;
                  break;
                }
                for (int mi2 = 0; mi2 <= nm && !fallback; ++mi2) {
                  uint32_t cs = mi2 == 0 ? chunk_start : marks[mi2 - 1] + 1;
                  uint32_t ce = mi2 == nm ? close_pos : marks[mi2];
                  if (cs >= ce) continue;
                  if (n >= WG_ENC_MAX_ITEMS) {
                    fallback = true;
                    break;
                  }
                  uint32_t span = ce - cs;
                  uint32_t icap = span + span / 4 + WG_ENC_ITEM_PAD;
                  if (acc + icap > scr_cap) {
                    fallback = true;
                    break;
                  }
                  s_start[n] = cs;
                  s_end[n] = ce;
                  s_ioff[n] = acc;
                  s_fidx[n] = m.field_start + fidx;
                  s_kind[n] = chunk_map ? 2 : 3;
                  ++n;
                  acc += icap;
                }
                if (fallback) break;
              }
            } else {
              if (!skip_value(c)) {
                fallback = true;
                break;
              }
              if (n >= WG_ENC_MAX_ITEMS) {
                fallback = true;
                break;
              }
              uint32_t span = c.pos - member_start;
              uint32_t icap = span + span / 4 + WG_ENC_ITEM_PAD;
              if (acc + icap > scr_cap) {
                fallback = true;
                break;
              }
              s_start[n] = member_start;
              s_end[n] = c.pos;
              s_ioff[n] = acc;
              s_fidx[n] = -1;
              s_kind[n] = 1;
              ++n;
              acc += icap;
            }
            // member separator
            skip_ws(c);
            uint8_t ch = peek(c);
            if (ch == ',') {
              c.pos++;
              continue;
            }
            if (ch == '}') {
              c.pos++;
              break;
            }
            fallback = true;
            break;
          }
        }
      } else {
        fallback = true;
      }
      if (fallback) mode_l = 0;
    }
    if (!lane) {
      s_res = r;
      s_res.pb_off = pb_off[req];
      s_mode = mode_l;
      s_nitems = n;
      s_next = 0;
      s_msg = msg_idx;
      s_args0 = args[0];
      s_args1 = args[1];
    }
  }
  __syncthreads();

  // ---- phase A' (s_mode == 2): multi-wave speculative structural scan ----
  // The serial member walk above streams ~64 KB through ONE wave (~12
  // us/KB post-SWAR) while 7 waves idle.  Here every wave summarizes
  // 256 B-granular windows of the arguments object under the 3 possible
  // entry states (out-of-string / in-string / in-string-with-pending-
  // escape: for VALID JSON backslashes only occur inside strings, so
  // exactly one global state assignment is consistent — anything
  // contradictory falls back to classic), one thread merges the
  // summaries into real per-window states/depths, a sparse second pass
  // collects depth-1/2 commas in document order, and members become
  // items IN PARALLEL (one wave per member).  Spans are constructed to
  // match the serial scanner byte-for-byte (trailing-ws-stripped plain
  // values; chunk edges at the same stride commas), so phases B-D and
  // the output wire are identical.  ANY anomaly -> s_mode 0 (classic
  // in-block path, exact classic error formats).  GGRMCP_MW_SCAN gates.
  if (s_mode == 2) {
    constexpr int MW_WIN = 256;
    constexpr int MW_D2 = 512;
    __shared__ int16_t mwdd[MW_WIN][3], mwdm[MW_WIN][3];
    __shared__ uint8_t mwend[MW_WIN][3], mwbad[MW_WIN][3];
    __shared__ uint8_t mwes[MW_WIN];
    __shared__ int16_t mwed[MW_WIN];
    __shared__ uint16_t mwc1[MW_WIN], mwc2[MW_WIN];
    __shared__ uint32_t mwp1[WG_ENC_MAX_ITEMS], mwp2[MW_D2];
    __shared__ uint32_t s_ob, s_close, s_winb;
    __shared__ int s_nwin, s_cwin, s_fb, s_n1, s_n2;
    __shared__ unsigned long long s_seenf;
    __shared__ uint32_t s_seeno;
    __shared__ uint16_t s_mcnt[WG_ENC_MAX_ITEMS];
    __shared__ uint32_t s_mcap[WG_ENC_MAX_ITEMS];
    __shared__ uint32_t s_mbase[WG_ENC_MAX_ITEMS];
    __shared__ uint32_t s_moff[WG_ENC_MAX_ITEMS];

#define MW_IS_WS(ch) ((ch) == ' ' || (ch) == '\t' || (ch) == '\n' || (ch) == '\r')

    // step 0: object opener, window geometry
    if (threadIdx.x == 0) {
      uint32_t p = s_args0;
      while (p < s_args1 && MW_IS_WS(src[p])) ++p;
      if (p >= s_args1 || src[p] != '{') {
        s_fb = 1;
      } else {
        s_ob = p;
        uint32_t span = s_args1 - (p + 1);
        uint32_t wb = (span + MW_WIN - 1) / MW_WIN;
        wb = (wb + 255u) & ~255u;
        if (wb == 0) wb = 256;
        s_winb = wb;
        s_nwin = (int)((span + wb - 1) / wb);
        s_fb = 0;
        s_cwin = -1;
        s_close = 0;
        s_n1 = 0;
        s_n2 = 0;
        s_seenf = 0;
        s_seeno = 0;
      }
    }
    __syncthreads();

    // pass 1: per-window summaries under all 3 entry states
    if (!s_fb) {
      const uint32_t base = s_ob + 1, aend = s_args1, wb = s_winb;
      for (int w = wave; w < s_nwin; w += WG_ENC_WAVES) {
        uint32_t wst = base + (uint32_t)w * wb;
        uint32_t wen = wst + wb;
        if (wen > aend) wen = aend;
        int st[3] = {0, 1, 1};
        uint32_t sk[3] = {0xFFFFFFFFu, 0xFFFFFFFFu, wst};
        int dd[3] = {0, 0, 0}, dm[3] = {0, 0, 0}, bad[3] = {0, 0, 0};
        for (uint32_t p = wst; p < wen; p += 4u * WAVE) {
          uint32_t off = p + 4u * (uint32_t)lane;
          uint32_t wd = load4_or(src, off, wen, 0);
          uint32_t hit = swar_eq(wd, '"') | swar_eq(wd, '\\') |
                         swar_eq(wd, '{') | swar_eq(wd, '}') |
                         swar_eq(wd, '[') | swar_eq(wd, ']') |
                         swar_eq(wd, ',');
          uint64_t lmask = __ballot(hit != 0);
          while (lmask) {
            int lf = __ffsll((long long)lmask) - 1;
            lmask &= lmask - 1;
            uint32_t lh = (uint32_t)__shfl(hit, lf, WAVE);
            uint32_t lw = (uint32_t)__shfl(wd, lf, WAVE);
            while (lh) {
              uint32_t bidx = (uint32_t)(__builtin_ctz(lh) >> 3);
              lh &= lh - 1;
              uint32_t pos = p + 4u * (uint32_t)lf + bidx;
              if (pos >= wen) {
                lh = 0;
                break;
              }
              uint8_t ch = (uint8_t)(lw >> (8 * bidx));
#pragma unroll
              for (int mi = 0; mi < 3; ++mi) {
                if (pos == sk[mi]) continue;
                if (st[mi]) {
                  if (ch == '\\')
                    sk[mi] = pos + 1;
                  else if (ch == '"')
                    st[mi] = 0;
                } else if (ch == '"') {
                  st[mi] = 1;
                } else if (ch == '{' || ch == '[') {
                  if (++dd[mi] > 30000) bad[mi] = 1;
                } else if (ch == '}' || ch == ']') {
                  if (--dd[mi] < dm[mi]) dm[mi] = dd[mi];
                  if (dd[mi] < -30000) bad[mi] = 1;
                } else if (ch == '\\') {
                  bad[mi] = 1;  // backslash outside any string
                }
              }
            }
          }
        }
        if (!lane) {
#pragma unroll
          for (int mi = 0; mi < 3; ++mi) {
            mwdd[w][mi] = (int16_t)dd[mi];
            mwdm[w][mi] = (int16_t)dm[mi];
            mwend[w][mi] = st[mi] ? (sk[mi] == wen ? 2 : 1) : 0;
            mwbad[w][mi] = (uint8_t)bad[mi];
          }
        }
      }
    }
    __syncthreads();

    // merge: resolve entry state + absolute depth per window, find the
    // window holding the object closer
    if (!s_fb && threadIdx.x == 0) {
      int stt = 0, dep = 1, cw = -1;
      for (int w = 0; w < s_nwin; ++w) {
        mwes[w] = (uint8_t)stt;
        mwed[w] = (int16_t)dep;
        if (mwbad[w][stt]) {
          s_fb = 2;
          break;
        }
        if (dep + (int)mwdm[w][stt] <= 0) {
          cw = w;
          break;
        }
        dep += (int)mwdd[w][stt];
        stt = (int)mwend[w][stt];
        if (dep > 30000) {
          s_fb = 2;
          break;
        }
      }
      if (!s_fb) {
        if (cw < 0)
          s_fb = 3;  // object never closes inside the span
        else
          s_cwin = cw;
      }
    }
    __syncthreads();

    // pass 2a: count depth-1 / depth-2 commas per qualifying window
    if (!s_fb) {
      const uint32_t base = s_ob + 1, wb = s_winb;
      for (int w = wave; w <= s_cwin; w += WG_ENC_WAVES) {
        int c1 = 0, c2 = 0;
        if ((int)mwed[w] + (int)mwdm[w][mwes[w]] <= 2) {
          uint32_t wst = base + (uint32_t)w * wb;
          uint32_t wen = wst + wb;
          if (wen > s_args1) wen = s_args1;
          int stt = (int)mwes[w], dep = (int)mwed[w];
          uint32_t sk = (stt == 2) ? wst : 0xFFFFFFFFu;
          if (stt == 2) stt = 1;
          for (uint32_t p = wst; p < wen && dep > 0; p += 4u * WAVE) {
            uint32_t off = p + 4u * (uint32_t)lane;
            uint32_t wd = load4_or(src, off, wen, 0);
            uint32_t hit = swar_eq(wd, '"') | swar_eq(wd, '\\') |
                           swar_eq(wd, '{') | swar_eq(wd, '}') |
                           swar_eq(wd, '[') | swar_eq(wd, ']') |
                           swar_eq(wd, ',');
            uint64_t lmask = __ballot(hit != 0);
            while (lmask && dep > 0) {
              int lf = __ffsll((long long)lmask) - 1;
              lmask &= lmask - 1;
              uint32_t lh = (uint32_t)__shfl(hit, lf, WAVE);
              uint32_t lw = (uint32_t)__shfl(wd, lf, WAVE);
              while (lh && dep > 0) {
                uint32_t bidx = (uint32_t)(__builtin_ctz(lh) >> 3);
                lh &= lh - 1;
                uint32_t pos = p + 4u * (uint32_t)lf + bidx;
                if (pos >= wen) {
                  lh = 0;
                  break;
                }
                uint8_t ch = (uint8_t)(lw >> (8 * bidx));
                if (pos == sk) continue;
                if (stt) {
                  if (ch == '\\')
                    sk = pos + 1;
                  else if (ch == '"')
                    stt = 0;
                } else if (ch == '"') {
                  stt = 1;
                } else if (ch == '{' || ch == '[') {
                  ++dep;
                } else if (ch == '}' || ch == ']') {
                  if (--dep == 0) {
                    if (ch != '}') {
                      atomicExch(&s_fb, 4);
                    } else if (!lane) {
                      s_close = pos;
                    }
                  }
                } else if (ch == ',') {
                  if (dep == 1)
                    ++c1;
                  else if (dep == 2)
                    ++c2;
                }
              }
            }
          }
        }
        if (!lane) {
          mwc1[w] = (uint16_t)c1;
          mwc2[w] = (uint16_t)c2;
        }
      }
    }
    __syncthreads();

    // pass 2b: exclusive scans -> per-window output bases; cap checks
    if (!s_fb && threadIdx.x == 0) {
      int a1 = 0, a2 = 0;
      for (int w = 0; w <= s_cwin; ++w) {
        int c1 = mwc1[w], c2 = mwc2[w];
        mwc1[w] = (uint16_t)a1;
        mwc2[w] = (uint16_t)a2;
        a1 += c1;
        a2 += c2;
      }
      s_n1 = a1;
      s_n2 = a2;
      if (a1 + 1 > WG_ENC_MAX_ITEMS || a2 > MW_D2) s_fb = 5;
    }
    __syncthreads();

    // pass 2c: place comma positions (window bases keep document order)
    if (!s_fb) {
      const uint32_t base = s_ob + 1, wb = s_winb;
      for (int w = wave; w <= s_cwin; w += WG_ENC_WAVES) {
        if ((int)mwed[w] + (int)mwdm[w][mwes[w]] > 2) continue;
        uint32_t wst = base + (uint32_t)w * wb;
        uint32_t wen = wst + wb;
        if (wen > s_args1) wen = s_args1;
        int stt = (int)mwes[w], dep = (int)mwed[w];
        uint32_t sk = (stt == 2) ? wst : 0xFFFFFFFFu;
        if (stt == 2) stt = 1;
        int o1 = mwc1[w], o2 = mwc2[w];
        for (uint32_t p = wst; p < wen && dep > 0; p += 4u * WAVE) {
          uint32_t off = p + 4u * (uint32_t)lane;
          uint32_t wd = load4_or(src, off, wen, 0);
          uint32_t hit = swar_eq(wd, '"') | swar_eq(wd, '\\') |
                         swar_eq(wd, '{') | swar_eq(wd, '}') |
                         swar_eq(wd, '[') | swar_eq(wd, ']') |
                         swar_eq(wd, ',');
          uint64_t lmask = __ballot(hit != 0);
          while (lmask && dep > 0) {
            int lf = __ffsll((long long)lmask) - 1;
            lmask &= lmask - 1;
            uint32_t lh = (uint32_t)__shfl(hit, lf, WAVE);
            uint32_t lw = (uint32_t)__shfl(wd, lf, WAVE);
            while (lh && dep > 0) {
              uint32_t bidx = (uint32_t)(__builtin_ctz(lh) >> 3);
              lh &= lh - 1;
              uint32_t pos = p + 4u * (uint32_t)lf + bidx;
              if (pos >= wen) {
                lh = 0;
                break;
              }
              uint8_t ch = (uint8_t)(lw >> (8 * bidx));
              if (pos == sk) continue;
              if (stt) {
                if (ch == '\\')
                  sk = pos + 1;
                else if (ch == '"')
                  stt = 0;
              } else if (ch == '"') {
                stt = 1;
              } else if (ch == '{' || ch == '[') {
                ++dep;
              } else if (ch == '}' || ch == ']') {
                --dep;
              } else if (ch == ',') {
                if (dep == 1) {
                  if (!lane) mwp1[o1] = pos;
                  ++o1;
                } else if (dep == 2) {
                  if (!lane) mwp2[o2] = pos;
                  ++o2;
                }
              }
            }
          }
        }
      }
    }
    __syncthreads();

    // members: M1 counts items + caps per member (validations + dup
    // detection), a serial scan assigns bases, M2 places the items.
    // Control flow between barriers uses a UNIFORM snapshot (s_go,
    // written by thread 0 before each barrier): raw s_fb reads race with
    // the atomic failure sets and could diverge the barrier counts.
    __shared__ int s_go;
    auto member_pass = [&](int place, int nmemb_l) {
      for (int m_i = wave; m_i < nmemb_l; m_i += WG_ENC_WAVES) {
        uint32_t mstart = (m_i == 0) ? s_ob + 1 : mwp1[m_i - 1] + 1;
        uint32_t mend = (m_i == nmemb_l - 1) ? s_close : mwp1[m_i];
        uint32_t kq = mstart;
        while (kq < mend && MW_IS_WS(src[kq])) ++kq;
        if (kq >= mend || src[kq] != '"') {
          atomicExch(&s_fb, 6);
          continue;
        }
        Ctx c;
        c.s = src;
        c.len = mend;
        c.pos = kq;
        c.out = nullptr;
        c.opos = 0;
        c.ocap = 0;
        c.t = t;
        c.lim = lim;
        c.status = E_OK;
        c.err_pos = 0;
        c.aux = 0;
        c.lane = lane;
        c.keybuf = keybufs[wave];
        uint32_t kst, krl;
        bool kesc;
        if (!string_span(c, &kst, &krl, &kesc) || kesc) {
          atomicExch(&s_fb, 6);
          continue;
        }
        uint64_t h = fnv1a64(c.s + kst, krl);
        const MsgEntry& mm = t.msgs[s_msg];
        int fidx = -1;
        for (int i = 0; i < mm.field_count; ++i) {
          const FieldEntry& fe = t.fields[mm.field_start + i];
          if ((fe.hash_json == h && fe.json_len == krl &&
               wave_equal(c, t.names + fe.json_off, c.s + kst, krl)) ||
              (fe.hash_orig == h && fe.name_len == krl &&
               wave_equal(c, t.names + fe.name_off, c.s + kst, krl))) {
            fidx = i;
            break;
          }
        }
        if (fidx < 0) {
          atomicExch(&s_fb, 6);
          continue;
        }
        const FieldEntry& fe = t.fields[mm.field_start + fidx];
        if (!place && !lane) {
          if (fidx < 64) {
            unsigned long long bit = 1ull << fidx;
            if (atomicOr(&s_seenf, bit) & bit) atomicExch(&s_fb, 6);
          }
          if (fe.flags & F_ONEOF) {
            uint32_t b = 1u << fe.oneof_id;
            if (atomicOr(&s_seeno, b) & b) atomicExch(&s_fb, 6);
          }
        }
        skip_ws(c);
        if (c.pos >= mend || src[c.pos] != ':') {
          atomicExch(&s_fb, 6);
          continue;
        }
        c.pos++;
        skip_ws(c);
        uint32_t vstart = c.pos;
        if (vstart >= mend) {
          atomicExch(&s_fb, 6);
          continue;
        }
        uint32_t vend = mend;
        while (vend > vstart && MW_IS_WS(src[vend - 1])) --vend;
        uint8_t vc = src[vstart];
        bool chunk_map = (fe.flags & F_MAP) && vc == '{';
        bool chunk_arr = (fe.flags & F_REPEATED) && !(fe.flags & F_MAP) &&
                         !is_packable(fe.kind) && vc == '[';
        bool big = (s_args1 - vstart) >= 2048;
        int cnt = 0;
        uint32_t cap_sum = 0;
        uint32_t ibase = place ? s_mbase[m_i] : 0;
        uint32_t ioff = place ? s_moff[m_i] : 0;
        if ((chunk_map || chunk_arr) && big) {
          uint8_t closer = chunk_map ? '}' : ']';
          if (vend <= vstart + 1 || src[vend - 1] != closer) {
            atomicExch(&s_fb, 6);
            continue;
          }
          uint32_t close_pos = vend - 1;
          uint32_t cstart = vstart + 1;
          while (cstart < close_pos && MW_IS_WS(src[cstart])) ++cstart;
          if (cstart < close_pos) {
            // stride marks over the collected depth-2 commas (same
            // cadence as scan_container(marks, cap=64, stride=4096))
            int lo2 = 0, hi2 = s_n2;
            while (lo2 < hi2) {
              int mid = (lo2 + hi2) >> 1;
              if (mwp2[mid] <= cstart)
                lo2 = mid + 1;
              else
                hi2 = mid;
            }
            uint32_t prev_cs = cstart, nmark = cstart + 4096;
            int nmk = 0;
            for (int q = lo2; q < s_n2 && nmk < 64; ++q) {
              uint32_t pos = mwp2[q];
              if (pos >= close_pos) break;
              if (pos >= nmark) {
                if (prev_cs < pos) {
                  uint32_t span = pos - prev_cs;
                  uint32_t icap = span + span / 4 + WG_ENC_ITEM_PAD;
                  if (place && !lane) {
                    s_start[ibase + cnt] = prev_cs;
                    s_end[ibase + cnt] = pos;
                    s_ioff[ibase + cnt] = ioff + cap_sum;
                    s_fidx[ibase + cnt] = mm.field_start + fidx;
                    s_kind[ibase + cnt] = chunk_map ? 2 : 3;
                  }
                  ++cnt;
                  cap_sum += icap;
                }
                prev_cs = pos + 1;
                nmark = pos + 4096;
                ++nmk;
              }
            }
            if (prev_cs < close_pos) {
              uint32_t span = close_pos - prev_cs;
              uint32_t icap = span + span / 4 + WG_ENC_ITEM_PAD;
              if (place && !lane) {
                s_start[ibase + cnt] = prev_cs;
                s_end[ibase + cnt] = close_pos;
                s_ioff[ibase + cnt] = ioff + cap_sum;
                s_fidx[ibase + cnt] = mm.field_start + fidx;
                s_kind[ibase + cnt] = chunk_map ? 2 : 3;
              }
              ++cnt;
              cap_sum += icap;
            }
          }
        } else {
          uint32_t span = vend - (kst - 1);
          uint32_t icap = span + span / 4 + WG_ENC_ITEM_PAD;
          if (place && !lane) {
            s_start[ibase] = kst - 1;
            s_end[ibase] = vend;
            s_ioff[ibase] = ioff;
            s_fidx[ibase] = -1;
            s_kind[ibase] = 1;
          }
          cnt = 1;
          cap_sum = icap;
        }
        if (!place && !lane) {
          s_mcnt[m_i] = (uint16_t)cnt;
          s_mcap[m_i] = cap_sum;
        }
      }
    };

    if (threadIdx.x == 0) s_go = s_fb;
    __syncthreads();
    int nmemb = 0;
    if (!s_go) {
      uint32_t p0 = s_ob + 1;
      while (p0 < s_close && MW_IS_WS(src[p0])) ++p0;
      nmemb = (p0 >= s_close) ? 0 : s_n1 + 1;
      member_pass(0, nmemb);
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      if (!s_fb) {
        uint32_t nb = 0, ob2 = 0;
        for (int m_i = 0; m_i < nmemb; ++m_i) {
          s_mbase[m_i] = nb;
          s_moff[m_i] = ob2;
          nb += s_mcnt[m_i];
          ob2 += s_mcap[m_i];
        }
        if (nb > (uint32_t)WG_ENC_MAX_ITEMS || ob2 > scr_cap)
          s_fb = 7;
        else
          s_nitems = (int)nb;
      }
      s_go = s_fb;
    }
    __syncthreads();
    if (!s_go) member_pass(1, nmemb);
    __syncthreads();
    if (threadIdx.x == 0) s_mode = s_fb ? 0 : 1;
    __syncthreads();
#undef MW_IS_WS
  }
  const int n_items = s_nitems;
  const int msg_idx = s_msg;

  // ---- phase B: encode items (dynamic wave grabs, bounded) ----------------
  if (s_mode == 1 && max_phase >= 1) {
    for (int guard = 0; guard <= WG_ENC_MAX_ITEMS + 1; ++guard) {
      int idx = 0;
      if (!lane) idx = atomicAdd(&s_next, 1);
      idx = __shfl(idx, 0, WAVE);
      if (idx >= n_items) break;
      uint32_t span = s_end[idx] - s_start[idx];
      Ctx c;
      c.s = src;
      c.len = s_end[idx];
      c.pos = s_start[idx];
      c.out = scr + s_ioff[idx];
      c.opos = 0;
      c.ocap = span + span / 4 + WG_ENC_ITEM_PAD;
      c.t = t;
      c.lim = lim;
      c.status = E_OK;
      c.err_pos = 0;
      c.aux = 0;
      c.lane = lane;
      c.keybuf = keybufs[wave];
      bool ok = encode_walk(c, msg_idx, (int)s_kind[idx], s_fidx[idx]);
      if (!ok || c.status != E_OK) {
        // classic reproduces exact error position/detail formats
        if (!lane) atomicCAS(&s_mode, 1, 0);
        continue;
      }
      if (!lane) s_olen[idx] = c.opos;
    }
  }
  __syncthreads();

  // ---- phase C: offsets (one thread) --------------------------------------
  if (s_mode == 1 && threadIdx.x == 0 && max_phase >= 2) {
    uint32_t off = 0;
    for (int i = 0; i < n_items; ++i) {
      s_foff[i] = off;
      off += s_olen[i];
    }
    if (off > pbcap) {
      s_mode = 0;  // shouldn't happen (caps per item), but never truncate
    } else {
      s_res.pb_len = off;
      s_next = 0;
    }
  }
  __syncthreads();

  // ---- phase D: compact items into the pb arena / classic fallback --------
  if (s_mode == 1 && max_phase >= 2) {
    for (int s = wave; s < n_items; s += WG_ENC_WAVES) {
      const uint8_t* p = scr + s_ioff[s];
      uint8_t* d = pbout + s_foff[s];
      uint32_t len = s_olen[s];
      for (uint32_t i = lane; i < len; i += WAVE) d[i] = p[i];
    }
  } else if (s_mode == 0 && wave == 0) {
    wg_enc_classic_one(src, src_len, pbout, pbcap, results,
                       id_slots + (size_t)req * ID_SLOT_BYTES, keybufs[0], t,
                       lim, req, lane);
    if (!lane) results[req].pb_off = pb_off[req];
  }
  __syncthreads();

  // ---- finalize ------------------------------------------------------------
  if (s_mode != 0 && threadIdx.x == 0) {
    if (s_mode == 1 && max_phase < 2) s_res.pb_len = 0;  // debug phases
    results[req] = s_res;
  }
}
#endif  // GGRMCP_HOST_SIM
