// k_pb2json — batched response kernel (gfx950, wave64).
//
// One 64-lane wave per request.  Replaces the reference gateway's CPU
// response path: protojson.Marshal of the gRPC response message
// (reflection.go:381) plus the JSON-RPC ToolCallResult envelope encode
// (handler.go:252-270, 290): protobuf wire walk -> JSON text in a scratch
// arena, then lane-parallel JSON-string escaping (exclusive wave scan of
// per-byte expansion via __shfl) into the final envelope
//   {"jsonrpc":"2.0","id":<id>,"result":{"content":[{"type":"text",
//    "text":"<escaped>"}],"isError":false}}.
//
// protojson output semantics: camelCase field names, 64-bit ints quoted,
// enums by name (unknown values by number), bytes as padded std base64,
// default-valued non-presence fields omitted, Timestamp/Duration RFC3339 /
// "Ns" forms with 3-digit-group nano trimming, wrappers unwrapped,
// Struct/Value/ListValue as raw JSON, Empty as {}.  Fields are emitted in
// wire order with adjacent-occurrence grouping for repeated/map fields;
// out-of-order wire (no real serializer produces it) and google.protobuf.Any
// flag E_UNSUPPORTED so the host transcodes that request (counted, never
// silent).  Doubles print as 17-significant-digit shortest-trimmed decimals
// (value-exact round trip; not always the minimal digit string).

#include "common.h"

// Mid-level emitters are deliberately NOT force-inlined: the iterative
// walker calls them at many sites, and inlining every put_double/dtoa17
// chain into the (already large) walker sent hipcc into a multi-minute
// optimizer blowup.  Inside each noinline function the tiny primitives
// (putc_/puts_/read_varint/put_escaped) still inline, so the per-byte hot
// loops keep their state in registers; the call overhead is per-field.
#ifdef GGRMCP_HOST_SIM
#define DEVN __attribute__((noinline))
#else
#define DEVN __device__ __noinline__
#endif

#ifndef WPB
#define WPB 4
#endif

struct DCtx {
  const uint8_t* pb;
  uint32_t len, pos;
  uint8_t* out;
  uint32_t opos, ocap;
  Tables t;
  int32_t status;
  int lane;
};

DEV bool dfail(DCtx& c, int32_t code) {
  if (c.status == E_OK) c.status = code;
  return false;
}

DEV bool putc_(DCtx& c, uint8_t ch) {
  if (c.opos >= c.ocap) return dfail(c, E_OVERFLOW);
  if (!c.lane) c.out[c.opos] = ch;
  c.opos++;
  return true;
}

DEV bool puts_(DCtx& c, const char* s, uint32_t n) {
  if (c.opos + n > c.ocap) return dfail(c, E_OVERFLOW);
  for (uint32_t i = c.lane; i < n; i += WAVE) c.out[c.opos + i] = (uint8_t)s[i];
  c.opos += n;
  return true;
}

DEV bool put_u64_dec(DCtx& c, uint64_t v) {
  if (c.opos + 20 > c.ocap) return dfail(c, E_OVERFLOW);
  uint32_t n = 1;
  {
    uint64_t x = v;
    while (x >= 10) {
      x /= 10;
      ++n;
    }
  }
  if (!c.lane) u64_to_dec(c.out + c.opos, v);
  c.opos += n;
  return true;
}

DEV bool put_i64_dec(DCtx& c, int64_t v) {
  if (v < 0) {
    if (!putc_(c, '-')) return false;
    return put_u64_dec(c, (uint64_t)(~v) + 1ull);
  }
  return put_u64_dec(c, (uint64_t)v);
}

// ---------------------------------------------------------------------------
// JSON string escaping — lane-parallel with wave prefix scan
// ---------------------------------------------------------------------------

DEV uint32_t esc_len(uint8_t b) {
  if (b == '"' || b == '\\') return 2;
  if (b == '\b' || b == '\f' || b == '\n' || b == '\r' || b == '\t') return 2;
  if (b < 0x20) return 6;  // \u00XX
  return 1;
}

// escape bytes packed little-endian into a register (no per-lane scratch
// array: an addressable tmp[6] forced a private-memory round trip per byte,
// measured at ~77 us per KB of string)
DEV uint64_t esc_pack(uint8_t b, uint32_t el) {
  if (el == 1) return b;
  const char* HEX = "0123456789abcdef";
  switch (b) {
    case '"': return 0x5cull | ((uint64_t)'"' << 8);
    case '\\': return ((uint64_t)'\\' << 8) | 0x5c;
    case '\b': return ((uint64_t)'b' << 8) | 0x5c;
    case '\f': return ((uint64_t)'f' << 8) | 0x5c;
    case '\n': return ((uint64_t)'n' << 8) | 0x5c;
    case '\r': return ((uint64_t)'r' << 8) | 0x5c;
    case '\t': return ((uint64_t)'t' << 8) | 0x5c;
    default:  // \u00XX
      return 0x5cull | ((uint64_t)'u' << 8) | ((uint64_t)'0' << 16) |
             ((uint64_t)'0' << 24) | ((uint64_t)(uint8_t)HEX[b >> 4] << 32) |
             ((uint64_t)(uint8_t)HEX[b & 15] << 40);
  }
}

// append src[0..n) JSON-escaped; lane-parallel with a no-escape fast path.
// Invalid UTF-8 (proto3 string fields must be valid; protojson errors) ->
// E_UNSUPPORTED so the host re-attempt surfaces the proper error.
// validate=false skips the UTF-8 re-check: used on SLICES of text the
// walker already validated (a slice boundary can split a multi-byte
// sequence, which would false-reject; escaping itself is per-byte safe)
DEV bool put_escaped(DCtx& c, const uint8_t* src, uint32_t n,
                     bool validate = true) {
  if (validate && !utf8_span_valid(src, n, c.lane))
    return dfail(c, E_UNSUPPORTED);
  uint32_t base = 0;
  while (base < n) {
    // 256 B super-window: escape-free spans (the overwhelmingly common
    // case) copy one dword per lane — the 64 B byte/lane loop was
    // memory-latency bound at ~26 us/KB
    uint32_t sw = n - base;
    if (sw > 4u * WAVE) sw = 4u * WAVE;
    uint32_t off = base + 4u * (uint32_t)c.lane;
    uint32_t w = load4_or(src, off, n, 'x');
    // needs-escape bytes: < 0x20, '"', '\\' (fill 'x' never hits)
    uint32_t hit = swar_eq(w, '"') | swar_eq(w, '\\') |
                   swar_zero(w & 0xE0E0E0E0u);
    if (!__ballot(hit != 0)) {
      if (c.opos + sw > c.ocap) return dfail(c, E_OVERFLOW);
      uint32_t lo = 4u * (uint32_t)c.lane;
      if (off + 4 <= n && lo + 4 <= sw) {
        __builtin_memcpy(c.out + c.opos + lo, &w, 4);
      } else {
        for (uint32_t j = 0; j < 4; ++j)
          if (lo + j < sw) c.out[c.opos + lo + j] = (uint8_t)(w >> (8 * j));
      }
      c.opos += sw;
      base += sw;
      continue;
    }
    // dirty super-window: original 64 B-window scan/expand path
    uint32_t end = base + sw;
    for (uint32_t b2 = base; b2 < end; b2 += WAVE) {
      uint32_t i = b2 + c.lane;
      uint32_t win = end - b2 < WAVE ? end - b2 : WAVE;
      uint8_t b = i < end ? src[i] : 'x';
      uint32_t el = i < end ? esc_len(b) : 0;
      uint64_t esc_mask = __ballot(el > 1);
      if (esc_mask == 0) {
        if (c.opos + win > c.ocap) return dfail(c, E_OVERFLOW);
        if (i < end) c.out[c.opos + c.lane] = b;
        c.opos += win;
        continue;
      }
      // inclusive wave scan of el
      uint32_t inc = el;
      #pragma unroll
      for (int d = 1; d < WAVE; d <<= 1) {
        uint32_t up = __shfl_up(inc, d, WAVE);
        if (c.lane >= d) inc += up;
      }
      uint32_t total = __shfl(inc, WAVE - 1, WAVE);
      if (c.opos + total > c.ocap) return dfail(c, E_OVERFLOW);
      if (i < end) {
        uint64_t bytes = esc_pack(b, el);
        uint32_t at = c.opos + inc - el;
        for (uint32_t k = 0; k < el; ++k)
          c.out[at + k] = (uint8_t)(bytes >> (8 * k));
      }
      c.opos += total;
    }
    base = end;
  }
  return true;
}

// append a quoted escaped string
DEV bool put_json_string(DCtx& c, const uint8_t* src, uint32_t n) {
  if (!putc_(c, '"')) return false;
  if (!put_escaped(c, src, n)) return false;
  return putc_(c, '"');
}

// ---------------------------------------------------------------------------
// base64 encode (bytes fields) — lane-parallel, 3 bytes/lane per window
// ---------------------------------------------------------------------------

DEV bool put_base64(DCtx& c, const uint8_t* src, uint32_t n) {
  static const char B64[] =
      "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789+/";
  uint32_t out_n = (n + 2) / 3 * 4;
  if (c.opos + out_n > c.ocap) return dfail(c, E_OVERFLOW);
  uint32_t groups = (n + 2) / 3;
  for (uint32_t g0 = 0; g0 < groups; g0 += WAVE) {
    uint32_t g = g0 + c.lane;
    if (g < groups) {
      uint32_t i = g * 3;
      uint32_t b0 = src[i];
      uint32_t b1 = i + 1 < n ? src[i + 1] : 0;
      uint32_t b2 = i + 2 < n ? src[i + 2] : 0;
      uint32_t v = (b0 << 16) | (b1 << 8) | b2;
      uint8_t* dst = c.out + c.opos + g * 4;
      dst[0] = B64[(v >> 18) & 63];
      dst[1] = B64[(v >> 12) & 63];
      dst[2] = i + 1 < n ? B64[(v >> 6) & 63] : '=';
      dst[3] = i + 2 < n ? B64[v & 63] : '=';
    }
  }
  c.opos += out_n;
  return true;
}

// ---------------------------------------------------------------------------
// double -> decimal text (17 significant digits, trailing zeros trimmed)
// ---------------------------------------------------------------------------

__constant__ double DPOW10[23] = {1e0,  1e1,  1e2,  1e3,  1e4,  1e5,  1e6,  1e7,
                                  1e8,  1e9,  1e10, 1e11, 1e12, 1e13, 1e14, 1e15,
                                  1e16, 1e17, 1e18, 1e19, 1e20, 1e21, 1e22};

// d * 10^e as a double-double (steps through exact powers)
DEV DD dd_scale10(double d, int e) {
  DD x{d, 0.0};
  while (e > 22) {
    x = dd_mul_exact(x, 1e22);
    e -= 22;
  }
  while (e < -22) {
    x = dd_div_exact(x, 1e22);
    e += 22;
  }
  if (e >= 0) return dd_mul_exact(x, DPOW10[e]);
  return dd_div_exact(x, DPOW10[-e]);
}

// digits * 10^(e10 - ndig + 1) ?= d, verified in double-double.  digits
// (< 1e17 < 2^57) splits exactly into hi + lo doubles.
DEV bool digits_roundtrip(uint64_t digits, int ndig, int e10, double d) {
#pragma clang fp contract(off) reassociate(off)
  double dh = (double)digits;                       // rounded
  double dl = (double)(int64_t)(digits - (uint64_t)(int64_t)dh);
  DD x{dh, dl};
  int e = e10 - ndig + 1;
  int ee = e;
  while (ee > 22) {
    x = dd_mul_exact(x, 1e22);
    ee -= 22;
  }
  while (ee < -22) {
    x = dd_div_exact(x, 1e22);
    ee += 22;
  }
  x = ee >= 0 ? dd_mul_exact(x, DPOW10[ee]) : dd_div_exact(x, DPOW10[-ee]);
  // round-trips iff the decimal lies within d's rounding interval: strictly
  // inside the half-gap on its side, or exactly ON the boundary when d's
  // mantissa is even (IEEE round-to-nearest-even resolves the tie to d —
  // e.g. 60982807099928340 parses to ...336, which repr prints at 16
  // digits; rejecting the tie forced a needless 17-digit form)
  double diff = (x.hi - d) + x.lo;
  bool above = diff >= 0;
  if (diff < 0) diff = -diff;
  // direction must be +-infinity: a finite direction constant below
  // DBL_MAX makes nextafter step DOWNWARD for d above it (negative gap,
  // false rejection near the top of the range — hypothesis-found with
  // the float variant near FLT_MAX)
  double gap_half = above ? (nextafter(d, HUGE_VAL) - d) * 0.5
                          : (d - nextafter(d, -HUGE_VAL)) * 0.5;
  if (diff < gap_half) return true;
  return diff == gap_half && (__builtin_bit_cast(uint64_t, d) & 1) == 0;
}

// float32 variant: does the decimal parse (to double, then to float — the
// protojson/strtof pipeline) land back on f?
DEV bool digits_roundtrip_f(uint64_t digits, int ndig, int e10, float f) {
#pragma clang fp contract(off) reassociate(off)
  double dh = (double)digits;
  double dl = (double)(int64_t)(digits - (uint64_t)(int64_t)dh);
  DD x{dh, dl};
  int ee = e10 - ndig + 1;
  while (ee > 22) {
    x = dd_mul_exact(x, 1e22);
    ee -= 22;
  }
  while (ee < -22) {
    x = dd_div_exact(x, 1e22);
    ee += 22;
  }
  x = ee >= 0 ? dd_mul_exact(x, DPOW10[ee]) : dd_div_exact(x, DPOW10[-ee]);
  double diff = (x.hi - (double)f) + x.lo;
  bool above = diff >= 0;
  if (diff < 0) diff = -diff;
  double gap_half = above
      ? ((double)nextafterf(f, HUGE_VALF) - (double)f) * 0.5
      : ((double)f - (double)nextafterf(f, -HUGE_VALF)) * 0.5;
  if (diff < gap_half) return true;
  return diff == gap_half && (__builtin_bit_cast(uint32_t, f) & 1) == 0;
}

// Generate `prec` significant decimal digits of d (>0, finite); returns the
// digit block and adjusts *e10_out.  Precision-search wrapper below picks the
// shortest precision whose parse-back round-trips — matching protojson's
// shortest-representation output.
// nearest integer with exact halves rounded to EVEN — shortest-round-trip
// printers (Ryu, Go strconv; protojson uses them) break the both-candidates-
// round-trip tie this way, e.g. float32 1048576.25 prints "1048576.2" not
// "1048576.3" (fuzz-found divergence)
DEV uint64_t dd_round_half_even(DD s) {
#pragma clang fp contract(off) reassociate(off)
  uint64_t dg = (uint64_t)s.hi;
  // (s.hi - dg) is exact (both < 2^58, within one ulp of each other).
  // For scaled values >= 2^53 the cast itself is quantized: s.hi is an
  // integral double with ulp up to 8, and the compensation term s.lo
  // carries the true value up to +-ulp/2 AWAY from that integer — so the
  // adjustment must walk, not step once (found: 17-digit blocks off by
  // 2-4 for scaled >= 4e16).
  double frac = (s.hi - (double)dg) + s.lo;
  while (frac < 0.0) {
    --dg;
    frac += 1.0;
  }
  while (frac >= 1.0) {
    ++dg;
    frac -= 1.0;
  }
  if (frac > 0.5 || (frac == 0.5 && (dg & 1))) ++dg;
  return dg;
}

DEV uint64_t gen_digits(double d, int prec, int* e10_out) {
  int e10 = (int)floor(log10(d));
  DD s = dd_scale10(d, (prec - 1) - e10);
  uint64_t digits = dd_round_half_even(s);
  uint64_t hi = 1;
  for (int i = 0; i < prec; ++i) hi *= 10;
  if (digits >= hi) {
    // rounded up across a decade (999..9.7 -> 100..00): the dropped digit
    // is zero, no re-round needed
    digits /= 10;
    e10 += 1;
  } else if (digits < hi / 10) {
    // log10 estimate was one high
    e10 -= 1;
    s = dd_scale10(d, (prec - 1) - e10);
    digits = dd_round_half_even(s);
    if (digits >= hi) {
      digits /= 10;
      e10 += 1;
    }
  }
  *e10_out = e10;
  return digits;
}

// emit d (>0, finite) as decimal text directly into the output (caller
// guarantees >= 40 bytes of cap).  Digits live in a u64 the whole time —
// the previous buffer-based form staged through scratch arrays (dig[17] +
// buf[32]) and cost ~9 us per double.
DEV void put_digit(DCtx& c, uint8_t ch) {
  if (!c.lane) c.out[c.opos] = ch;
  c.opos++;
}

DEVN bool emit_double_body(DCtx& c, double d, bool as_float) {
  // integral fast path (exact digits).  For float fields only below 2^24:
  // protojson prints the SHORTEST digits that round-trip as float32.
  if (d == trunc(d) &&
      (as_float ? d < 16777216.0 : d < 9.007199254740992e15)) {
    uint32_t n = 1;
    uint64_t v = (uint64_t)d;
    while (n < 20 && v >= DEC_P10[n]) ++n;
    for (uint32_t i = 0; i < n; ++i)
      put_digit(c, (uint8_t)('0' + (uint32_t)((v / DEC_P10[n - 1 - i]) % 10)));
    return true;
  }
  // shortest precision whose parse-back round-trips
  int p_lo = as_float ? 6 : 15;
  int p_hi = as_float ? 9 : 17;
  int e10 = 0;
  uint64_t digits = 0;
  int prec = p_hi;
  for (int p = p_lo; p <= p_hi; ++p) {
    int e;
    uint64_t dg = gen_digits(d, p, &e);
    bool ok = as_float ? digits_roundtrip_f(dg, p, e, (float)d)
                       : digits_roundtrip(dg, p, e, d);
    if (ok || p == p_hi) {
      digits = dg;
      e10 = e;
      prec = p;
      break;
    }
  }
  // trim trailing zeros (integer form)
  int ndig = prec;
  while (ndig > 1 && digits % 10 == 0) {
    digits /= 10;
    --ndig;
  }
  // digit k (0-based, MSB first) of the ndig-digit block
#define DIGIT_AT(k) ((uint8_t)('0' + (uint32_t)((digits / DEC_P10[ndig - 1 - (k)]) % 10)))
  if (e10 >= -6 && e10 <= 20) {
    if (e10 >= 0) {
      int ip = e10 + 1;  // digits before the point
      for (int i = 0; i < ip; ++i)
        put_digit(c, i < ndig ? DIGIT_AT(i) : (uint8_t)'0');
      if (ndig > ip) {
        put_digit(c, '.');
        for (int i = ip; i < ndig; ++i) put_digit(c, DIGIT_AT(i));
      }
    } else {
      put_digit(c, '0');
      put_digit(c, '.');
      for (int i = 0; i < -e10 - 1; ++i) put_digit(c, '0');
      for (int i = 0; i < ndig; ++i) put_digit(c, DIGIT_AT(i));
    }
  } else {
    put_digit(c, DIGIT_AT(0));
    if (ndig > 1) {
      put_digit(c, '.');
      for (int i = 1; i < ndig; ++i) put_digit(c, DIGIT_AT(i));
    }
    put_digit(c, 'e');
    int e = e10;
    if (e < 0) {
      put_digit(c, '-');
      e = -e;
    } else {
      put_digit(c, '+');
    }
    if (e >= 100) {
      put_digit(c, (uint8_t)('0' + e / 100));
      e %= 100;
      put_digit(c, (uint8_t)('0' + e / 10));
      put_digit(c, (uint8_t)('0' + e % 10));
    } else {
      put_digit(c, (uint8_t)('0' + e / 10));
      put_digit(c, (uint8_t)('0' + e % 10));
    }
  }
#undef DIGIT_AT
  return true;
}

DEVN bool put_double(DCtx& c, double d, bool as_float) {
  if (as_float) d = (double)(float)d;
  if (d != d) return puts_(c, "\"NaN\"", 5);
  if (isinf(d))
    return d > 0 ? puts_(c, "\"Infinity\"", 10) : puts_(c, "\"-Infinity\"", 11);
  // outermost decade + subnormal range: composed base-10 scaling carries a
  // few ulp of error, enough to print digits that strtod rounds to inf (or
  // to mis-round subnormals).  Host transcodes these slots (counted).
  {
    double ad = d < 0 ? -d : d;
    if (ad > 1e308 || (ad != 0.0 && ad < 1e-306))
      return dfail(c, E_UNSUPPORTED);
  }
  if (c.opos + 40 > c.ocap) return dfail(c, E_OVERFLOW);
  if (d == 0.0) {
    if (signbit(d)) put_digit(c, '-');
    put_digit(c, '0');
    return true;
  }
  if (d < 0) {
    put_digit(c, '-');
    d = -d;
  }
  return emit_double_body(c, d, as_float);
}

// ---------------------------------------------------------------------------
// Timestamp / Duration formatting
// ---------------------------------------------------------------------------

DEV void civil_from_days(int64_t z, int64_t* y, int* m, int* d) {
  z += 719468;
  int64_t era = (z >= 0 ? z : z - 146096) / 146097;
  uint64_t doe = (uint64_t)(z - era * 146097);
  uint64_t yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
  int64_t yr = (int64_t)yoe + era * 400;
  uint64_t doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
  uint64_t mp = (5 * doy + 2) / 153;
  uint64_t dd = doy - (153 * mp + 2) / 5 + 1;
  uint64_t mm = mp + (mp < 10 ? 3 : (uint64_t)-9);
  *y = yr + (mm <= 2);
  *m = (int)mm;
  *d = (int)dd;
}

DEV uint32_t put2(uint8_t* o, int v) {
  o[0] = (uint8_t)('0' + v / 10);
  o[1] = (uint8_t)('0' + v % 10);
  return 2;
}

// nanos -> ".fff[fff[fff]]" trimmed in 3-digit groups (protojson rule)
DEV uint32_t put_nanos(uint8_t* o, int32_t nanos) {
  if (nanos == 0) return 0;
  uint32_t n = 0;
  o[n++] = '.';
  int digits = 9;
  if (nanos % 1000000 == 0) {
    digits = 3;
    nanos /= 1000000;
  } else if (nanos % 1000 == 0) {
    digits = 6;
    nanos /= 1000;
  }
  for (int i = digits - 1; i >= 0; --i) {
    o[n + i] = (uint8_t)('0' + nanos % 10);
    nanos /= 10;
  }
  return n + digits;
}

DEVN bool put_timestamp(DCtx& c, int64_t secs, int32_t nanos) {
  if (c.opos + 40 > c.ocap) return dfail(c, E_OVERFLOW);
  uint8_t buf[40];
  uint32_t n = 0;
  int64_t days = secs >= 0 ? secs / 86400 : (secs - 86399) / 86400;
  int64_t rem = secs - days * 86400;
  int64_t y;
  int mo, dd;
  civil_from_days(days, &y, &mo, &dd);
  buf[n++] = '"';
  buf[n] = (uint8_t)('0' + (y / 1000) % 10);
  buf[n + 1] = (uint8_t)('0' + (y / 100) % 10);
  buf[n + 2] = (uint8_t)('0' + (y / 10) % 10);
  buf[n + 3] = (uint8_t)('0' + y % 10);
  n += 4;
  buf[n++] = '-';
  n += put2(buf + n, mo);
  buf[n++] = '-';
  n += put2(buf + n, dd);
  buf[n++] = 'T';
  n += put2(buf + n, (int)(rem / 3600));
  buf[n++] = ':';
  n += put2(buf + n, (int)((rem / 60) % 60));
  buf[n++] = ':';
  n += put2(buf + n, (int)(rem % 60));
  n += put_nanos(buf + n, nanos);
  buf[n++] = 'Z';
  buf[n++] = '"';
  if (!c.lane)
    for (uint32_t i = 0; i < n; ++i) c.out[c.opos + i] = buf[i];
  c.opos += n;
  return true;
}

DEVN bool put_duration(DCtx& c, int64_t secs, int32_t nanos) {
  if (c.opos + 40 > c.ocap) return dfail(c, E_OVERFLOW);
  uint8_t buf[40];
  uint32_t n = 0;
  buf[n++] = '"';
  bool neg = secs < 0 || nanos < 0;
  if (neg) buf[n++] = '-';
  uint64_t s = secs < 0 ? (uint64_t)(-secs) : (uint64_t)secs;
  int32_t ns = nanos < 0 ? -nanos : nanos;
  n += u64_to_dec(buf + n, s);
  n += put_nanos(buf + n, ns);
  buf[n++] = 's';
  buf[n++] = '"';
  if (!c.lane)
    for (uint32_t i = 0; i < n; ++i) c.out[c.opos + i] = buf[i];
  c.opos += n;
  return true;
}

// ---------------------------------------------------------------------------
// wire helpers
// ---------------------------------------------------------------------------

DEV bool read_varint(DCtx& c, uint64_t* v) {
  if (!get_varint(c.pb, c.len, &c.pos, v)) return dfail(c, E_PARSE);
  return true;
}

DEV bool read_fixed32(DCtx& c, uint32_t* v) {
  if (c.pos + 4 > c.len) return dfail(c, E_PARSE);
  *v = (uint32_t)c.pb[c.pos] | ((uint32_t)c.pb[c.pos + 1] << 8) |
       ((uint32_t)c.pb[c.pos + 2] << 16) | ((uint32_t)c.pb[c.pos + 3] << 24);
  c.pos += 4;
  return true;
}

DEV bool read_fixed64(DCtx& c, uint64_t* v) {
  if (c.pos + 8 > c.len) return dfail(c, E_PARSE);
  uint64_t r = 0;
  for (int i = 0; i < 8; ++i) r |= (uint64_t)c.pb[c.pos + i] << (8 * i);
  *v = r;
  c.pos += 8;
  return true;
}

DEV bool skip_wire(DCtx& c, uint32_t wt) {
  uint64_t v;
  switch (wt) {
    case W_VARINT: return read_varint(c, &v);
    case W_I64:
      if (c.pos + 8 > c.len) return dfail(c, E_PARSE);
      c.pos += 8;
      return true;
    case W_LEN: {
      if (!read_varint(c, &v)) return false;
      if (c.pos + v > c.len) return dfail(c, E_PARSE);
      c.pos += (uint32_t)v;
      return true;
    }
    case W_I32:
      if (c.pos + 4 > c.len) return dfail(c, E_PARSE);
      c.pos += 4;
      return true;
  }
  return dfail(c, E_PARSE);
}

DEV const FieldEntry* find_field(DCtx& c, const MsgEntry& m, uint32_t number) {
  // fields are sorted by number (tables.py).  Densely-numbered messages
  // (number == position+1, the overwhelmingly common schema shape) hit in
  // O(1); otherwise binary search.  The linear scan this replaces cost
  // ~300 cycles per field on 64-field messages (~150 ns/field measured).
  if (number >= 1 && number <= (uint32_t)m.field_count) {
    const FieldEntry& f = c.t.fields[m.field_start + number - 1];
    if (f.number == number) return &f;
  }
  int lo = 0, hi = m.field_count - 1;
  while (lo <= hi) {
    int mid = (lo + hi) >> 1;
    const FieldEntry& f = c.t.fields[m.field_start + mid];
    if (f.number == number) return &f;
    if (f.number < number) lo = mid + 1;
    else hi = mid - 1;
  }
  return nullptr;
}

DEV uint32_t expected_wire(const FieldEntry& f) {
  switch (f.kind) {
    case K_DOUBLE: case K_FIXED64: case K_SFIXED64: return W_I64;
    case K_FLOAT: case K_FIXED32: case K_SFIXED32: return W_I32;
    case K_STRING: case K_BYTES: case K_MESSAGE: return W_LEN;
    default: return W_VARINT;
  }
}

// ---------------------------------------------------------------------------
// decoder
// ---------------------------------------------------------------------------

// ---------------------------------------------------------------------------
// decoder — ITERATIVE walker.
//
// The first implementation recursed through nested messages; the compiler
// then had to keep the DCtx (and every frame) in scratch memory, so each
// processed byte paid several private-memory round trips (~170 cycles/byte
// measured).  This walker is a single flat function with an explicit frame
// stack: the context never has its address escape, every helper inlines,
// and the hot state lives in VGPRs.  It also removes the dynamic device
// stack entirely for this kernel.
// ---------------------------------------------------------------------------

// emit ONE scalar/string/bytes/enum value from the wire (no name, not
// K_MESSAGE — nested messages are the walker's job)
DEVN bool emit_scalar_value(DCtx& c, const FieldEntry& f) {
  uint64_t v;
  uint32_t v32;
  switch (f.kind) {
    case K_DOUBLE:
      if (!read_fixed64(c, &v)) return false;
      return put_double(c, __builtin_bit_cast(double, v), false);
    case K_FLOAT:
      if (!read_fixed32(c, &v32)) return false;
      return put_double(c, (double)__builtin_bit_cast(float, v32), true);
    case K_INT64:
    case K_SFIXED64: {
      int64_t sv;
      if (f.kind == K_INT64) {
        if (!read_varint(c, &v)) return false;
        sv = (int64_t)v;
      } else {
        if (!read_fixed64(c, &v)) return false;
        sv = (int64_t)v;
      }
      if (!putc_(c, '"')) return false;
      if (!put_i64_dec(c, sv)) return false;
      return putc_(c, '"');
    }
    case K_SINT64: {
      if (!read_varint(c, &v)) return false;
      if (!putc_(c, '"')) return false;
      if (!put_i64_dec(c, unzigzag64(v))) return false;
      return putc_(c, '"');
    }
    case K_UINT64:
    case K_FIXED64: {
      if (f.kind == K_UINT64) {
        if (!read_varint(c, &v)) return false;
      } else {
        if (!read_fixed64(c, &v)) return false;
      }
      if (!putc_(c, '"')) return false;
      if (!put_u64_dec(c, v)) return false;
      return putc_(c, '"');
    }
    case K_INT32: {
      if (!read_varint(c, &v)) return false;
      return put_i64_dec(c, (int64_t)(int32_t)(uint32_t)v);
    }
    case K_SINT32: {
      if (!read_varint(c, &v)) return false;
      return put_i64_dec(c, (int64_t)(int32_t)((uint32_t)(v >> 1) ^ (uint32_t)(0 - (v & 1))));
    }
    case K_SFIXED32: {
      if (!read_fixed32(c, &v32)) return false;
      return put_i64_dec(c, (int64_t)(int32_t)v32);
    }
    case K_UINT32: {
      if (!read_varint(c, &v)) return false;
      return put_u64_dec(c, (uint32_t)v);
    }
    case K_FIXED32: {
      if (!read_fixed32(c, &v32)) return false;
      return put_u64_dec(c, v32);
    }
    case K_BOOL: {
      if (!read_varint(c, &v)) return false;
      return v ? puts_(c, "true", 4) : puts_(c, "false", 5);
    }
    case K_ENUM: {
      if (!read_varint(c, &v)) return false;
      int32_t num = (int32_t)(int64_t)v;
      const EnumEntry& ee = c.t.enums[f.sub_index];
      for (int i = 0; i < ee.val_count; ++i) {
        const EnumValueEntry& ev = c.t.enum_vals[ee.val_start + i];
        if (ev.number == num)
          return put_json_string(c, c.t.names + ev.name_off, ev.name_len);
      }
      return put_i64_dec(c, num);  // unknown enum value -> number (protojson)
    }
    case K_STRING: {
      if (!read_varint(c, &v)) return false;
      if (c.pos + v > c.len) return dfail(c, E_PARSE);
      if (!put_json_string(c, c.pb + c.pos, (uint32_t)v)) return false;
      c.pos += (uint32_t)v;
      return true;
    }
    case K_BYTES: {
      if (!read_varint(c, &v)) return false;
      if (c.pos + v > c.len) return dfail(c, E_PARSE);
      if (!putc_(c, '"')) return false;
      if (!put_base64(c, c.pb + c.pos, (uint32_t)v)) return false;
      c.pos += (uint32_t)v;
      return putc_(c, '"');
    }
  }
  return dfail(c, E_UNSUPPORTED);
}

// packed array payload (numeric/enum/bool kinds only per protobuf)
DEVN bool emit_packed(DCtx& c, const FieldEntry& f, uint32_t end, bool* first) {
  while (c.pos < end) {
    if (!*first) {
      if (!putc_(c, ',')) return false;
    }
    *first = false;
    if (!emit_scalar_value(c, f)) return false;
  }
  return c.pos == end || dfail(c, E_PARSE);
}

// map key bytes -> JSON object key (always quoted)
DEVN bool emit_map_key(DCtx& c, const FieldEntry& kf, uint32_t key_pos,
                      bool have_key) {
  if (!putc_(c, '"')) return false;
  if (have_key) {
    uint32_t save = c.pos;
    c.pos = key_pos;
    uint64_t v;
    bool ok = true;
    switch (kf.kind) {
      case K_STRING: {
        ok = read_varint(c, &v);
        if (ok && c.pos + v > c.len) ok = dfail(c, E_PARSE);
        if (ok) ok = put_escaped(c, c.pb + c.pos, (uint32_t)v);
        break;
      }
      case K_BOOL: {
        ok = read_varint(c, &v);
        if (ok) ok = v ? puts_(c, "true", 4) : puts_(c, "false", 5);
        break;
      }
      case K_SINT64:
      case K_SINT32: {
        ok = read_varint(c, &v);
        if (ok) ok = put_i64_dec(c, unzigzag64(v));
        break;
      }
      case K_UINT64:
      case K_UINT32: {
        ok = read_varint(c, &v);
        if (ok) ok = put_u64_dec(c, v);
        break;
      }
      case K_FIXED64: {
        ok = read_fixed64(c, &v);
        if (ok) ok = put_u64_dec(c, v);
        break;
      }
      case K_SFIXED64: {
        ok = read_fixed64(c, &v);
        if (ok) ok = put_i64_dec(c, (int64_t)v);
        break;
      }
      case K_FIXED32: {
        uint32_t v32;
        ok = read_fixed32(c, &v32);
        if (ok) ok = put_u64_dec(c, v32);
        break;
      }
      case K_SFIXED32: {
        uint32_t v32;
        ok = read_fixed32(c, &v32);
        if (ok) ok = put_i64_dec(c, (int32_t)v32);
        break;
      }
      default: {  // int32/int64
        ok = read_varint(c, &v);
        if (ok) ok = put_i64_dec(c, (int64_t)v);
      }
    }
    c.pos = save;
    if (!ok) return false;
  } else {
    if (kf.kind != K_STRING && !putc_(c, '0')) return false;
  }
  if (!putc_(c, '"')) return false;
  return putc_(c, ':');
}

// default JSON for a missing map VALUE half
DEVN bool emit_map_default_value(DCtx& c, const FieldEntry& vf) {
  switch (vf.kind) {
    case K_STRING:
    case K_BYTES: return puts_(c, "\"\"", 2);
    case K_BOOL: return puts_(c, "false", 5);
    case K_MESSAGE: return puts_(c, "{}", 2);
    case K_INT64: case K_UINT64: case K_SINT64: case K_FIXED64:
    case K_SFIXED64:
      return puts_(c, "\"0\"", 3);
    case K_ENUM: {
      const EnumEntry& ee = c.t.enums[vf.sub_index];
      for (int i = 0; i < ee.val_count; ++i) {
        const EnumValueEntry& ev = c.t.enum_vals[ee.val_start + i];
        if (ev.number == 0)
          return put_json_string(c, c.t.names + ev.name_off, ev.name_len);
      }
      return putc_(c, '0');
    }
    default:
      return putc_(c, '0');
  }
}

// ---- scalar WKT leaves (no nesting) ----------------------------------------

DEVN bool decode_ts_dur_body(DCtx& c, uint32_t end, bool is_ts) {
  int64_t secs = 0;
  int32_t nanos = 0;
  while (c.pos < end) {
    uint64_t tag;
    if (!read_varint(c, &tag)) return false;
    uint32_t num = (uint32_t)(tag >> 3);
    if (num == 1 && (tag & 7) == W_VARINT) {
      uint64_t v;
      if (!read_varint(c, &v)) return false;
      secs = (int64_t)v;
    } else if (num == 2 && (tag & 7) == W_VARINT) {
      uint64_t v;
      if (!read_varint(c, &v)) return false;
      nanos = (int32_t)(int64_t)v;
    } else {
      if (!skip_wire(c, (uint32_t)(tag & 7))) return false;
    }
  }
  return is_ts ? put_timestamp(c, secs, nanos) : put_duration(c, secs, nanos);
}

DEVN bool decode_wrapper_body(DCtx& c, const MsgEntry& m, uint32_t end) {
  const FieldEntry& inner = c.t.fields[m.field_start];
  bool emitted = false;
  while (c.pos < end) {
    uint64_t tag;
    if (!read_varint(c, &tag)) return false;
    if ((uint32_t)(tag >> 3) == 1) {
      if (!emit_scalar_value(c, inner)) return false;
      emitted = true;
    } else {
      if (!skip_wire(c, (uint32_t)(tag & 7))) return false;
    }
  }
  if (!emitted) {
    switch (inner.kind) {
      case K_STRING: case K_BYTES: return puts_(c, "\"\"", 2);
      case K_BOOL: return puts_(c, "false", 5);
      case K_DOUBLE: case K_FLOAT: return putc_(c, '0');
      case K_INT64: case K_UINT64: return puts_(c, "\"0\"", 3);
      default: return putc_(c, '0');
    }
  }
  return true;
}

DEVN bool decode_fieldmask_body(DCtx& c, uint32_t end) {
  if (!putc_(c, '"')) return false;
  bool first = true;
  while (c.pos < end) {
    uint64_t tag;
    if (!read_varint(c, &tag)) return false;
    if ((uint32_t)(tag >> 3) != 1 || (tag & 7) != W_LEN) {
      if (!skip_wire(c, (uint32_t)(tag & 7))) return false;
      continue;
    }
    uint64_t slen;
    if (!read_varint(c, &slen)) return false;
    if (c.pos + slen > c.len) return dfail(c, E_PARSE);
    if (!first && !putc_(c, ',')) return false;
    first = false;
    bool up = false;  // snake -> camel
    for (uint32_t i = 0; i < (uint32_t)slen; ++i) {
      uint8_t ch = c.pb[c.pos + i];
      if (ch == '_') {
        up = true;
        continue;
      }
      if (up && ch >= 'a' && ch <= 'z') ch -= 32;
      up = false;
      if (!putc_(c, ch)) return false;
    }
    c.pos += (uint32_t)slen;
  }
  return putc_(c, '"');
}

// ---- the walker ------------------------------------------------------------

enum : uint8_t { FM_BODY = 0, FM_STRUCT = 1, FM_LIST = 2 };
enum : uint8_t { CK_NONE = 0, CK_ARRAY = 1, CK_MAP = 2 };

struct DFrame {
  uint32_t end;           // wire end of this body
  uint32_t prev_number;   // ascending-field check (FM_BODY)
  uint32_t cont_end;      // current map-entry end (CK_MAP with pushed value)
  int32_t msg_idx;
  int32_t cont_field;     // field-table index of the active container
  uint32_t cont_num;      // field number of the active container
  uint8_t mode;
  uint8_t first_member;
  uint8_t cont_kind;
  uint8_t cont_first;
};

DEVN bool map_entry_step(DCtx& c, DFrame& f, uint32_t eend, DFrame* stack,
                        int& sp, int* pushed);


// classify field f's message body starting at c.pos with length vlen.
// Scalar WKTs are emitted inline (returns 1=done), container bodies are
// described for a push (returns 2) with *push_mode set; error -> 0.
DEVN int enter_body(DCtx& c, int msg_idx, uint32_t vend, uint8_t* push_mode,
                   int32_t* push_idx, uint32_t* body_end) {
  *push_idx = msg_idx;
  *body_end = vend;
  const MsgEntry& m = c.t.msgs[msg_idx];
  switch (m.wkt_kind) {
    case WKT_TIMESTAMP:
      return decode_ts_dur_body(c, vend, true) ? 1 : 0;
    case WKT_DURATION:
      return decode_ts_dur_body(c, vend, false) ? 1 : 0;
    case WKT_WRAPPER:
      return decode_wrapper_body(c, m, vend) ? 1 : 0;
    case WKT_FIELDMASK:
      return decode_fieldmask_body(c, vend) ? 1 : 0;
    case WKT_ANY:
      return dfail(c, E_UNSUPPORTED) ? 1 : 0;
    case WKT_VALUE: {
      // one-shot oneof; nested struct/list push frames
      if (c.pos >= vend) return puts_(c, "null", 4) ? 1 : 0;
      uint64_t tag;
      if (!read_varint(c, &tag)) return 0;
      uint32_t num = (uint32_t)(tag >> 3);
      uint64_t v;
      switch (num) {
        case 1:
          if (!read_varint(c, &v)) return 0;
          return puts_(c, "null", 4) ? 1 : 0;
        case 2:
          if (!read_fixed64(c, &v)) return 0;
          return put_double(c, __builtin_bit_cast(double, v), false) ? 1 : 0;
        case 3: {
          if (!read_varint(c, &v)) return 0;
          if (c.pos + v > c.len) return dfail(c, E_PARSE) ? 1 : 0;
          if (!put_json_string(c, c.pb + c.pos, (uint32_t)v)) return 0;
          c.pos += (uint32_t)v;
          return 1;
        }
        case 4:
          if (!read_varint(c, &v)) return 0;
          return (v ? puts_(c, "true", 4) : puts_(c, "false", 5)) ? 1 : 0;
        case 5:
        case 6: {
          if (!read_varint(c, &v)) return 0;
          if (c.pos + v > c.len) return dfail(c, E_PARSE) ? 1 : 0;
          const FieldEntry* sf = find_field(c, m, num);
          if (!sf) return dfail(c, E_UNSUPPORTED) ? 1 : 0;
          // unwrap the nested Struct/ListValue HERE: re-entering through a
          // forced-inline recursive call sends the optimizer into runaway
          // inlining (hipcc never finished compiling)
          const MsgEntry& sm = c.t.msgs[sf->sub_index];
          *push_mode = (sm.wkt_kind == WKT_LISTVALUE) ? FM_LIST : FM_STRUCT;
          *push_idx = sf->sub_index;
          *body_end = c.pos + (uint32_t)v;
          return 2;
        }
      }
      return dfail(c, E_PARSE) ? 1 : 0;
    }
    case WKT_STRUCT:
      *push_mode = FM_STRUCT;
      return 2;
    case WKT_LISTVALUE:
      *push_mode = FM_LIST;
      return 2;
    default:
      *push_mode = FM_BODY;
      return 2;
  }
}


DEVN bool map_entry_step(DCtx& c, DFrame& f, uint32_t eend, DFrame* stack,
                        int& sp, int* pushed) {
  *pushed = 0;
  const MsgEntry& em = c.t.msgs[c.t.fields[f.cont_field].sub_index];
  const FieldEntry& kf = c.t.fields[em.field_start];
  const FieldEntry& vf = c.t.fields[em.field_start + 1];
  // fast path: the canonical wire order every real serializer emits —
  // key (field 1) then value (field 2).  ALL structural checks run before
  // any byte is emitted, so falling back to the general two-pass walk
  // never double-emits.
  if (c.pos < eend) {
    uint32_t save0 = c.pos;
    bool committed = false;
    uint64_t tag;
    if (!read_varint(c, &tag)) return false;
    if ((tag >> 3) == 1 && (uint32_t)(tag & 7) == expected_wire(kf)) {
      uint32_t kp = c.pos;
      if (!skip_wire(c, (uint32_t)(tag & 7))) return false;
      if (c.pos < eend) {
        uint64_t vtag;
        if (!read_varint(c, &vtag)) return false;
        if ((vtag >> 3) == 2) {
          if (vf.kind == K_MESSAGE) {
            uint64_t v;
            if (!read_varint(c, &v)) return false;
            if (c.pos + v > c.len) return dfail(c, E_PARSE);
            uint32_t vend = c.pos + (uint32_t)v;
            if (vend == eend) {
              // commit: emit key, then enter/push the message value
              uint32_t body_pos = c.pos;
              if (!emit_map_key(c, kf, kp, true)) return false;
              c.pos = body_pos;
              uint8_t pm = FM_BODY;
              int32_t pidx = vf.sub_index;
              uint32_t bend = vend;
              int r = enter_body(c, vf.sub_index, vend, &pm, &pidx, &bend);
              if (r == 0) return false;
              if (r == 1) {
                c.pos = eend;
                return true;
              }
              if (sp >= MAX_RECURSE) return dfail(c, E_LIMIT);
              f.cont_end = eend;
              DFrame& nf = stack[sp++];
              nf.end = bend;
              nf.prev_number = 0;
              nf.cont_end = 0;
              nf.msg_idx = pidx;
              nf.cont_field = -1;
              nf.cont_num = 0;
              nf.mode = pm;
              nf.first_member = 1;
              nf.cont_kind = CK_NONE;
              nf.cont_first = 1;
              if (!putc_(c, pm == FM_LIST ? '[' : '{')) return false;
              *pushed = 1;
              return true;
            }
          } else if ((uint32_t)(vtag & 7) == expected_wire(vf)) {
            // commit: key, value, then skip any trailing entry fields
            uint32_t vpos = c.pos;
            if (!emit_map_key(c, kf, kp, true)) return false;
            c.pos = vpos;
            if (!emit_scalar_value(c, vf)) return false;
            while (c.pos < eend) {
              uint64_t t2;
              if (!read_varint(c, &t2)) return false;
              if (!skip_wire(c, (uint32_t)(t2 & 7))) return false;
            }
            committed = true;
          }
        }
      }
    }
    if (committed) return true;
    c.pos = save0;
  }
  bool have_key = false, have_val = false;
  uint32_t key_pos = 0, val_pos = 0;
  while (c.pos < eend) {
    uint64_t tag;
    if (!read_varint(c, &tag)) return false;
    uint32_t num = (uint32_t)(tag >> 3);
    if (num == 1) {
      have_key = true;
      key_pos = c.pos;
    } else if (num == 2) {
      have_val = true;
      val_pos = c.pos;
    }
    if (!skip_wire(c, (uint32_t)(tag & 7))) return false;
  }
  if (!emit_map_key(c, kf, key_pos, have_key)) return false;
  if (!have_val) {
    if (!emit_map_default_value(c, vf)) return false;
    c.pos = eend;
    return true;
  }
  c.pos = val_pos;
  if (vf.kind == K_MESSAGE) {
    uint64_t v;
    if (!read_varint(c, &v)) return false;
    if (c.pos + v > c.len) return dfail(c, E_PARSE);
    uint32_t vend = c.pos + (uint32_t)v;
    uint8_t pm = FM_BODY;
    int32_t pidx = vf.sub_index;
    uint32_t bend = vend;
    int r = enter_body(c, vf.sub_index, vend, &pm, &pidx, &bend);
    if (r == 0) return false;
    if (r == 1) {
      c.pos = eend;
      return true;
    }
    if (sp >= MAX_RECURSE) return dfail(c, E_LIMIT);
    f.cont_end = eend;  // resume point after the child pops
    DFrame& nf = stack[sp++];
    nf.end = bend;
    nf.prev_number = 0;
    nf.cont_end = 0;
    nf.msg_idx = pidx;
    nf.cont_field = -1;
    nf.cont_num = 0;
    nf.mode = pm;
    nf.first_member = 1;
    nf.cont_kind = CK_NONE;
    nf.cont_first = 1;
    if (!putc_(c, pm == FM_LIST ? '[' : '{')) return false;
    *pushed = 1;
    return true;
  }
  if (!emit_scalar_value(c, vf)) return false;
  c.pos = eend;
  return true;
}

// bare_top: the top frame emits NO surrounding braces — used by the
// workgroup-cooperative kernel, whose per-item walks produce the comma-
// joined members of an enclosing object the caller writes itself.
DEV bool decode_walk(DCtx& c, int top_msg_idx, uint32_t top_end,
                     bool bare_top = false) {
  DFrame stack[MAX_RECURSE];
  int sp = 0;

  // enter the top-level body (responses are plain messages; transcode tests
  // may decode WKT-typed messages directly)
  if (bare_top) {
    // known-plain body (the wg scanner rejects WKT tops): push the frame
    // directly, no opening brace
    DFrame& f = stack[sp++];
    f.end = top_end;
    f.prev_number = 0;
    f.cont_end = 0;
    f.msg_idx = top_msg_idx;
    f.cont_field = -1;
    f.cont_num = 0;
    f.mode = FM_BODY;
    f.first_member = 1;
    f.cont_kind = CK_NONE;
    f.cont_first = 1;
  } else {
    uint8_t pm = FM_BODY;
    int32_t pidx = top_msg_idx;
    uint32_t bend = top_end;
    int r = enter_body(c, top_msg_idx, top_end, &pm, &pidx, &bend);
    if (r == 0) return false;
    if (r == 1) return c.status == E_OK;
    DFrame& f = stack[sp++];
    f.end = bend;
    f.prev_number = 0;
    f.cont_end = 0;
    f.msg_idx = pidx;
    f.cont_field = -1;
    f.cont_num = 0;
    f.mode = pm;
    f.first_member = 1;
    f.cont_kind = CK_NONE;
    f.cont_first = 1;
    if (!putc_(c, pm == FM_LIST ? '[' : '{')) return false;
  }

  while (sp > 0) {
    DFrame& f = stack[sp - 1];

    // -- resume an active container after a child pop / entry finish ------
    if (f.cont_kind == CK_ARRAY) {
      bool more = false;
      uint32_t wt = 0;
      if (c.pos < f.end) {
        uint32_t save = c.pos;
        uint64_t ntag;
        if (!read_varint(c, &ntag)) return false;
        if ((uint32_t)(ntag >> 3) == f.cont_num) {
          more = true;
          wt = (uint32_t)(ntag & 7);
        } else {
          c.pos = save;
        }
      }
      if (!more) {
        if (!putc_(c, ']')) return false;
        f.cont_kind = CK_NONE;
        continue;
      }
      const FieldEntry& fe = c.t.fields[f.cont_field];
      if (!putc_(c, ',')) return false;
      if (wt == W_LEN && expected_wire(fe) != W_LEN) {
        uint64_t plen;
        if (!read_varint(c, &plen)) return false;
        uint32_t pend = c.pos + (uint32_t)plen;
        if (pend > c.len) return dfail(c, E_PARSE);
        bool first = false;  // comma already emitted? emit_packed prepends
        // emit_packed writes separators before each elem when !first; we
        // already wrote one comma, so mark first=true for the first elem
        first = true;
        if (!emit_packed(c, fe, pend, &first)) return false;
        continue;
      }
      if (fe.kind == K_MESSAGE) {
        uint64_t v;
        if (!read_varint(c, &v)) return false;
        if (c.pos + v > c.len) return dfail(c, E_PARSE);
        uint32_t vend = c.pos + (uint32_t)v;
        uint8_t pm = FM_BODY;
        int32_t pidx = fe.sub_index;
        uint32_t bend = vend;
        int r = enter_body(c, fe.sub_index, vend, &pm, &pidx, &bend);
        if (r == 0) return false;
        if (r == 1) {
          c.pos = vend;
          continue;
        }
        if (sp >= MAX_RECURSE) return dfail(c, E_LIMIT);
        DFrame& nf = stack[sp++];
        nf.end = bend;
        nf.prev_number = 0;
        nf.cont_end = 0;
        nf.msg_idx = pidx;
        nf.cont_field = -1;
        nf.cont_num = 0;
        nf.mode = pm;
        nf.first_member = 1;
        nf.cont_kind = CK_NONE;
        nf.cont_first = 1;
        if (!putc_(c, pm == FM_LIST ? '[' : '{')) return false;
        continue;
      }
      if (wt != expected_wire(fe)) return dfail(c, E_PARSE);
      if (!emit_scalar_value(c, fe)) return false;
      continue;
    }
    if (f.cont_kind == CK_MAP) {
      // a pushed map VALUE just finished: jump to the entry end, then check
      // for an adjacent entry of the same field
      if (f.cont_end) {
        c.pos = f.cont_end;
        f.cont_end = 0;
      }
      bool more = false;
      if (c.pos < f.end) {
        uint32_t save = c.pos;
        uint64_t ntag;
        if (!read_varint(c, &ntag)) return false;
        if ((uint32_t)(ntag >> 3) == f.cont_num) {
          more = true;
        } else {
          c.pos = save;
        }
      }
      if (!more) {
        // a Struct's entries ARE its body: the body step emits the closing
        // brace when pos reaches end (emitting here double-closed it)
        if (f.mode != FM_STRUCT && !putc_(c, '}')) return false;
        f.cont_kind = CK_NONE;
        continue;
      }
      if (!putc_(c, ',')) return false;
      // start the next entry
      uint64_t elen;
      if (!read_varint(c, &elen)) return false;
      uint32_t eend = c.pos + (uint32_t)elen;
      if (eend > c.len) return dfail(c, E_PARSE);
      int pushed = 0;
      if (!map_entry_step(c, f, eend, stack, sp, &pushed)) return false;
      continue;
    }

    // -- body loops ------------------------------------------------------
    if (f.mode == FM_LIST) {
      if (c.pos >= f.end) {
        if (!putc_(c, ']')) return false;
        --sp;
        continue;
      }
      uint64_t tag;
      if (!read_varint(c, &tag)) return false;
      if ((uint32_t)(tag >> 3) != 1 || (tag & 7) != W_LEN) {
        if (!skip_wire(c, (uint32_t)(tag & 7))) return false;
        continue;
      }
      uint64_t elen;
      if (!read_varint(c, &elen)) return false;
      uint32_t eend = c.pos + (uint32_t)elen;
      if (eend > c.len) return dfail(c, E_PARSE);
      if (!f.first_member && !putc_(c, ',')) return false;
      f.first_member = 0;
      const MsgEntry& m = c.t.msgs[f.msg_idx];
      const FieldEntry& vf = c.t.fields[m.field_start];
      uint8_t pm = FM_BODY;
      int32_t pidx = vf.sub_index;
      uint32_t bend = eend;
      int r = enter_body(c, vf.sub_index, eend, &pm, &pidx, &bend);
      if (r == 0) return false;
      if (r == 1) {
        c.pos = eend;
        continue;
      }
      if (sp >= MAX_RECURSE) return dfail(c, E_LIMIT);
      DFrame& nf = stack[sp++];
      nf.end = bend;
      nf.prev_number = 0;
      nf.cont_end = 0;
      nf.msg_idx = pidx;
      nf.cont_field = -1;
      nf.cont_num = 0;
      nf.mode = pm;
      nf.first_member = 1;
      nf.cont_kind = CK_NONE;
      nf.cont_first = 1;
      if (!putc_(c, pm == FM_LIST ? '[' : '{')) return false;
      continue;
    }
    if (f.mode == FM_STRUCT) {
      // map<string, Value> on field 1 rendered as a bare object
      if (c.pos >= f.end) {
        if (!putc_(c, '}')) return false;
        --sp;
        continue;
      }
      uint64_t tag;
      if (!read_varint(c, &tag)) return false;
      if ((uint32_t)(tag >> 3) != 1 || (tag & 7) != W_LEN) {
        if (!skip_wire(c, (uint32_t)(tag & 7))) return false;
        continue;
      }
      uint64_t elen;
      if (!read_varint(c, &elen)) return false;
      uint32_t eend = c.pos + (uint32_t)elen;
      if (eend > c.len) return dfail(c, E_PARSE);
      if (!f.first_member && !putc_(c, ',')) return false;
      f.first_member = 0;
      const MsgEntry& m = c.t.msgs[f.msg_idx];
      f.cont_field = m.field_start;  // entries field of the struct map
      f.cont_num = 1;
      f.cont_kind = CK_MAP;  // struct entries use map machinery, sans '{'
      int pushed = 0;
      if (!map_entry_step(c, f, eend, stack, sp, &pushed)) return false;
      continue;
    }

    // FM_BODY
    if (c.pos >= f.end) {
      if (!(bare_top && sp == 1) && !putc_(c, '}')) return false;
      --sp;
      continue;
    }
    uint64_t tag;
    if (!read_varint(c, &tag)) return false;
    uint32_t num = (uint32_t)(tag >> 3), wt = (uint32_t)(tag & 7);
    const MsgEntry& m = c.t.msgs[f.msg_idx];
    const FieldEntry* fe = find_field(c, m, num);
    if (fe == nullptr) {
      if (!skip_wire(c, wt)) return false;  // unknown field: dropped
      continue;
    }
    // out-of-order / non-adjacent duplicates -> host transcodes
    if (num <= f.prev_number) return dfail(c, E_UNSUPPORTED);
    f.prev_number = num;
    bool repeated = (fe->flags & F_REPEATED) != 0;
    bool is_map = (fe->flags & F_MAP) != 0;
    // proto3 default-omission for singular non-presence fields
    if (!repeated && !(fe->flags & F_HAS_PRESENCE)) {
      if (wt == W_VARINT) {
        uint32_t save = c.pos;
        uint64_t v;
        if (!read_varint(c, &v)) return false;
        if (v == 0) continue;
        c.pos = save;
      } else if (wt == W_LEN && (fe->kind == K_STRING || fe->kind == K_BYTES)) {
        uint32_t save = c.pos;
        uint64_t v;
        if (!read_varint(c, &v)) return false;
        if (v == 0) continue;
        c.pos = save;
      } else if (wt == W_I64 || wt == W_I32) {
        uint32_t save = c.pos;
        uint64_t v = 1;
        if (wt == W_I64) {
          if (!read_fixed64(c, &v)) return false;
        } else {
          uint32_t v32;
          if (!read_fixed32(c, &v32)) return false;
          v = v32;
        }
        if (v == 0) continue;
        c.pos = save;
      }
    }
    if (!f.first_member && !putc_(c, ',')) return false;
    f.first_member = 0;
    if (!putc_(c, '"')) return false;
    if (!puts_(c, (const char*)(c.t.names + fe->json_off), fe->json_len))
      return false;
    if (!putc_(c, '"')) return false;
    if (!putc_(c, ':')) return false;

    if (is_map) {
      if (!putc_(c, '{')) return false;
      uint64_t elen;
      if (!read_varint(c, &elen)) return false;
      uint32_t eend = c.pos + (uint32_t)elen;
      if (eend > c.len) return dfail(c, E_PARSE);
      f.cont_field = (int32_t)(fe - c.t.fields);
      f.cont_num = num;
      f.cont_kind = CK_MAP;
      int pushed = 0;
      if (!map_entry_step(c, f, eend, stack, sp, &pushed)) return false;
      continue;
    }
    if (repeated) {
      if (!putc_(c, '[')) return false;
      f.cont_field = (int32_t)(fe - c.t.fields);
      f.cont_num = num;
      f.cont_kind = CK_ARRAY;
      // first element (wt already read)
      if (wt == W_LEN && expected_wire(*fe) != W_LEN) {
        uint64_t plen;
        if (!read_varint(c, &plen)) return false;
        uint32_t pend = c.pos + (uint32_t)plen;
        if (pend > c.len) return dfail(c, E_PARSE);
        bool first = true;
        if (!emit_packed(c, *fe, pend, &first)) return false;
        continue;
      }
      if (fe->kind == K_MESSAGE) {
        uint64_t v;
        if (!read_varint(c, &v)) return false;
        if (c.pos + v > c.len) return dfail(c, E_PARSE);
        uint32_t vend = c.pos + (uint32_t)v;
        uint8_t pm = FM_BODY;
        int32_t pidx = fe->sub_index;
        uint32_t bend = vend;
        int r = enter_body(c, fe->sub_index, vend, &pm, &pidx, &bend);
        if (r == 0) return false;
        if (r == 1) {
          c.pos = vend;
          continue;
        }
        if (sp >= MAX_RECURSE) return dfail(c, E_LIMIT);
        DFrame& nf = stack[sp++];
        nf.end = bend;
        nf.prev_number = 0;
        nf.cont_end = 0;
        nf.msg_idx = pidx;
        nf.cont_field = -1;
        nf.cont_num = 0;
        nf.mode = pm;
        nf.first_member = 1;
        nf.cont_kind = CK_NONE;
        nf.cont_first = 1;
        if (!putc_(c, pm == FM_LIST ? '[' : '{')) return false;
        continue;
      }
      if (wt != expected_wire(*fe)) return dfail(c, E_PARSE);
      if (!emit_scalar_value(c, *fe)) return false;
      continue;
    }
    // singular message or scalar
    if (fe->kind == K_MESSAGE) {
      if (wt != W_LEN) return dfail(c, E_PARSE);
      uint64_t v;
      if (!read_varint(c, &v)) return false;
      if (c.pos + v > c.len) return dfail(c, E_PARSE);
      uint32_t vend = c.pos + (uint32_t)v;
      uint8_t pm = FM_BODY;
      int32_t pidx = fe->sub_index;
      uint32_t bend = vend;
      int r = enter_body(c, fe->sub_index, vend, &pm, &pidx, &bend);
      if (r == 0) return false;
      if (r == 1) {
        c.pos = vend;
        continue;
      }
      if (sp >= MAX_RECURSE) return dfail(c, E_LIMIT);
      DFrame& nf = stack[sp++];
      nf.end = bend;
      nf.prev_number = 0;
      nf.cont_end = 0;
      nf.msg_idx = pidx;
      nf.cont_field = -1;
      nf.cont_num = 0;
      nf.mode = pm;
      nf.first_member = 1;
      nf.cont_kind = CK_NONE;
      nf.cont_first = 1;
      if (!putc_(c, pm == FM_LIST ? '[' : '{')) return false;
      continue;
    }
    if (wt != expected_wire(*fe)) return dfail(c, E_PARSE);
    if (!emit_scalar_value(c, *fe)) return false;
  }
  return c.status == E_OK;
}

// ---------------------------------------------------------------------------
// kernel
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(WPB * WAVE) k_pb2json(
    const uint8_t* __restrict__ resp_bytes, const uint32_t* __restrict__ resp_off,
    const int32_t* __restrict__ msg_idx_arr,
    const uint8_t* __restrict__ id_slots, const SlotResult* __restrict__ enc_results,
    uint8_t* __restrict__ scratch, const uint32_t* __restrict__ scratch_off,
    uint8_t* __restrict__ final_out, const uint32_t* __restrict__ final_off,
    DecodeResult* __restrict__ results, const int32_t* __restrict__ skip,
    Tables t, int n_req, int mode) {
  int wave_in_block = threadIdx.x / WAVE;
  int lane = lane_id();
  for (int req = blockIdx.x * WPB + wave_in_block; req < n_req;
       req += gridDim.x * WPB) {
    DecodeResult r;
    r.status = E_OK;
    r.out_off = final_off[req];
    r.out_len = 0;
    r.pad = 0;
    if (skip && skip[req]) {
      // host handles this slot (error / fallback); leave empty
      if (!lane) results[req] = r;
      continue;
    }
    // ---- phase 1: decode protobuf -> JSON in scratch ----
    DCtx c;
    c.pb = resp_bytes + resp_off[req];
    c.len = resp_off[req + 1] - resp_off[req];
    c.pos = 0;
    c.out = scratch + scratch_off[req];
    c.opos = 0;
    c.ocap = scratch_off[req + 1] - scratch_off[req];
    c.t = t;
    c.status = E_OK;
    c.lane = lane;
    decode_walk(c, msg_idx_arr[req], c.len);
    if (c.status != E_OK) {
      r.status = c.status;
      if (!lane) results[req] = r;
      continue;
    }
    uint32_t json_len = c.opos;

    // ---- phase 2: output ----
    DCtx o;
    o.pb = nullptr;
    o.len = 0;
    o.pos = 0;
    o.out = final_out + final_off[req];
    o.opos = 0;
    o.ocap = final_off[req + 1] - final_off[req];
    o.t = t;
    o.status = E_OK;
    o.lane = lane;
    if (mode == 1) {
      // bare JSON (transcode tests): copy scratch out
      if (json_len > o.ocap) {
        r.status = E_OVERFLOW;
        if (!lane) results[req] = r;
        continue;
      }
      for (uint32_t i = lane; i < json_len; i += WAVE)
        o.out[i] = scratch[scratch_off[req] + i];
      o.opos = json_len;
    } else if (mode == 2) {
      // streaming content item: the chunk pre-wrapped + escaped so the host
      // assembles a multi-chunk ToolCallResult by joining byte slices
      // (json-escaping 4096 chunk texts per stream in Python dominated the
      // streaming step otherwise)
      static const char CP1[] = "{\"type\":\"text\",\"text\":\"";
      static const char CP2[] = "\"}";
      bool ok = puts_(o, CP1, sizeof(CP1) - 1);
      if (ok) ok = put_escaped(o, scratch + scratch_off[req], json_len);
      if (ok) ok = puts_(o, CP2, sizeof(CP2) - 1);
      if (!ok) {
        r.status = o.status;
        if (!lane) results[req] = r;
        continue;
      }
    } else {
      // JSON-RPC result envelope (handler.go:265-270 + TextContent wrap)
      static const char P1[] = "{\"jsonrpc\":\"2.0\",\"id\":";
      static const char P2[] =
          ",\"result\":{\"content\":[{\"type\":\"text\",\"text\":\"";
      static const char P3[] = "\"}],\"isError\":false}}";
      bool ok = puts_(o, P1, sizeof(P1) - 1);
      uint32_t idl = enc_results ? enc_results[req].id_len : 0;
      if (ok && idl) {
        if (o.opos + idl > o.ocap) ok = dfail(o, E_OVERFLOW);
        if (ok) {
          const uint8_t* idp = id_slots + (size_t)req * ID_SLOT_BYTES;
          for (uint32_t i = lane; i < idl; i += WAVE) o.out[o.opos + i] = idp[i];
          o.opos += idl;
        }
      } else if (ok) {
        ok = puts_(o, "null", 4);
      }
      if (ok) ok = puts_(o, P2, sizeof(P2) - 1);
      if (ok) ok = put_escaped(o, scratch + scratch_off[req], json_len);
      if (ok) ok = puts_(o, P3, sizeof(P3) - 1);
      if (!ok) {
        r.status = o.status;
        if (!lane) results[req] = r;
        continue;
      }
    }
    r.status = E_OK;
    r.out_len = o.opos;
    if (!lane) results[req] = r;
  }
}

// Gather each slot's produced bytes into a tight, contiguous output buffer
// so the D2H copy moves only used bytes (the final arena is provisioned for
// worst-case 16x expansion; copying its full capacity back dominated decode
// time for large payloads: 268 MB vs ~25 MB used at 256 x 64 KB).
extern "C" __global__ void k_compact_out(
    const uint8_t* __restrict__ src, const uint32_t* __restrict__ src_off,
    const DecodeResult* __restrict__ results, uint8_t* __restrict__ dst,
    uint32_t dst_cap, int n_req) {
  // destination offsets come from results[].out_off, which k_tight_scan
  // rewrote to the packed layout; dst_cap guards the (rare) case where
  // the packed total exceeds the staging arena — the host then takes the
  // arena-copy fallback and this kernel must not write out of bounds
  int req = blockIdx.x;
  if (req >= n_req) return;
  uint32_t len = results[req].out_len;
  uint32_t doff = results[req].out_off;
  if (doff + len > dst_cap || doff + len < doff) return;
  const uint8_t* s = src + src_off[req];
  uint8_t* d = dst + doff;
  uint32_t t = threadIdx.x, stride = blockDim.x;
  // dword-wide main copy (both offsets 4-aligned by construction)
  uint32_t words = len >> 2;
  const uint32_t* s4 = (const uint32_t*)s;
  uint32_t* d4 = (uint32_t*)d;
  for (uint32_t i = t; i < words; i += stride) d4[i] = s4[i];
  for (uint32_t i = (words << 2) + t; i < len; i += stride) d[i] = s[i];
}

#ifndef GGRMCP_HOST_SIM
// Rewrites results[].out_off from arena offsets to a PACKED (4-aligned)
// layout entirely on device, so the host never round-trips between the
// decode kernels and k_compact_out (the old flow synced, prefix-summed
// out_lens on the host, and uploaded a tight-offset table — one extra
// blocking sync + one extra H2D per batch; the copy-op count is the
// engine-aggregate ceiling, profiles/contention_r02.md).  One block;
// contiguous per-thread chunks then a Hillis-Steele scan of the 256
// partials in LDS.  The packed total lands in rs[n-1] (host recomputes
// it from out_off + out_len); if it exceeds the arena the host restores
// arena offsets from final_off and takes the fallback copy.
extern "C" __global__ void __launch_bounds__(256) k_tight_scan(
    DecodeResult* __restrict__ rs, int n) {
  __shared__ uint32_t partial[256];
  int t = threadIdx.x;
  int per = (n + 255) / 256;
  int lo = t * per, hi = n < lo + per ? n : lo + per;
  uint32_t sum = 0;
  for (int i = lo; i < hi; ++i) sum += (rs[i].out_len + 3u) & ~3u;
  partial[t] = sum;
  __syncthreads();
  for (int d = 1; d < 256; d <<= 1) {
    uint32_t v = (t >= d) ? partial[t - d] : 0;
    __syncthreads();
    partial[t] += v;
    __syncthreads();
  }
  uint32_t acc = (t == 0) ? 0 : partial[t - 1];
  for (int i = lo; i < hi; ++i) {
    uint32_t len4 = (rs[i].out_len + 3u) & ~3u;
    rs[i].out_off = acc;
    acc += len4;
  }
}
#endif  // GGRMCP_HOST_SIM

#ifndef GGRMCP_HOST_SIM
// ---------------------------------------------------------------------------
// k_pb2json_wg — workgroup-per-request decode for LARGE responses
// (BASELINE config 3: 64 KB payloads).
//
// The per-wave walker serializes one request through 64 lanes at a
// ~26 us/KB floor (a 64 KB response costs ~1.7 ms of wave latency and the
// whole batch inherits it).  Here one WORKGROUP (WG_DEC_WAVES wave64s)
// owns one request:
//   A. one thread scans the top-level wire for contiguous field runs
//      ("items") — O(#fields), just varint/tag walking;
//   B. waves grab items dynamically and decode each item with the SAME
//      decode_walk machinery (bare_top: no surrounding braces) into a
//      private scratch region, then compute its escaped length;
//   C. one thread prefix-sums the escaped item lengths into final
//      offsets (comma-joined, skipping default-omitted empty items);
//   D. wave 0 writes the JSON-RPC envelope framing while every wave
//      escape-copies its items into place.
// Requests the scanner cannot split safely (WKT-typed top message,
// out-of-order top fields, > WG_DEC_MAX_ITEMS runs) fall back to the
// classic single-wave mode-0 path inside the same block, preserving
// exact semantics.  The engine routes only mode-0 slots with
// wire_len >= WG_DEC_MIN_BYTES here (skip tag == 2).
// ---------------------------------------------------------------------------

static __device__ const char WG_P1[] = "{\"jsonrpc\":\"2.0\",\"id\":";
static __device__ const char WG_P2[] =
    ",\"result\":{\"content\":[{\"type\":\"text\",\"text\":\"";
static __device__ const char WG_P3[] = "\"}],\"isError\":false}}";

extern "C" __global__ void __launch_bounds__(WG_DEC_WAVES * WAVE) k_pb2json_wg(
    const uint8_t* __restrict__ resp_bytes,
    const uint32_t* __restrict__ resp_off,
    const int32_t* __restrict__ msg_idx_arr,
    const uint8_t* __restrict__ id_slots,
    const SlotResult* __restrict__ enc_results, uint8_t* __restrict__ scratch,
    const uint32_t* __restrict__ scratch_off, uint8_t* __restrict__ final_out,
    const uint32_t* __restrict__ final_off, DecodeResult* __restrict__ results,
    const int32_t* __restrict__ skip, Tables t, int n_req, int max_phase) {
  // max_phase: debug bisection knob (GGRMCP_WG_PHASES; 3 = full pipeline)
  int req = blockIdx.x;
  if (req >= n_req) return;
  if (!skip || skip[req] != 2) return;  // 2 = routed to this kernel

  // items = top-level field runs; slices = fixed-size spans of the items'
  // produced text, so the escape passes load-balance across waves even
  // when one map/array run carries most of the payload
  constexpr int MAX_SLICES = 1024;
  __shared__ uint32_t s_start[WG_DEC_MAX_ITEMS + 1];
  __shared__ uint32_t s_outlen[WG_DEC_MAX_ITEMS];
  // map / repeated runs are CHUNKED at entry boundaries so one big run
  // doesn't serialize on a single wave; each chunk's walk emits the full
  // `"name":{...}` / `"name":[...]` syntax and the joiner trims
  // s_head bytes of leading name syntax and s_tail closing brackets
  __shared__ uint16_t s_head[WG_DEC_MAX_ITEMS];
  __shared__ uint8_t s_tail[WG_DEC_MAX_ITEMS];
  __shared__ uint16_t s_sl_item[MAX_SLICES];
  __shared__ uint32_t s_sl_off[MAX_SLICES];   // offset within the item text
  __shared__ uint32_t s_sl_esc[MAX_SLICES];   // escaped length of the slice
  __shared__ uint32_t s_sl_fin[MAX_SLICES];   // final-buffer offset
  __shared__ uint8_t s_sl_comma[MAX_SLICES];  // prepend ',' (item starts)
  __shared__ int s_nitems, s_nslices, s_next, s_err, s_mode;
  __shared__ uint32_t s_total, s_slice_bytes;

  const uint8_t* pb = resp_bytes + resp_off[req];
  const uint32_t wire_len = resp_off[req + 1] - resp_off[req];
  uint8_t* scr = scratch + scratch_off[req];
  const uint32_t scr_cap = scratch_off[req + 1] - scratch_off[req];
  uint8_t* fout = final_out + final_off[req];
  const uint32_t fcap = final_off[req + 1] - final_off[req];
  const int lane = lane_id();
  const int wave = threadIdx.x / WAVE;
  const int msg_idx = msg_idx_arr[req];

  // ---- phase A: top-level field-run scan (one thread) ---------------------
  // Runs group contiguous same-number tags; MAP / REPEATED runs split into
  // ~4 KB chunks at entry boundaries (chunk walks re-emit the name syntax;
  // s_head/s_tail trim it at join time) so a payload-dominating map field
  // parallelizes across waves instead of serializing on one.
  if (threadIdx.x == 0) {
    s_next = 0;
    s_err = E_OK;
    s_mode = 1;
    int n = 0;
    uint32_t pos = 0, prev_num = 0, cur = 0xFFFFFFFFu;
    uint32_t chunk_head = 0;   // head trim for continuation chunks of cur
    bool cur_split = false;    // cur run may chunk at entry boundaries
    uint32_t chunk_wire = 0;   // wire bytes accumulated in the open chunk
    const MsgEntry& tm = t.msgs[msg_idx];
    bool ok = tm.wkt_kind == WKT_NONE;
    while (ok && pos < wire_len) {
      uint32_t tag_start = pos;
      uint64_t tag;
      if (!get_varint(pb, wire_len, &pos, &tag)) { ok = false; break; }
      uint32_t num = (uint32_t)(tag >> 3), wt = (uint32_t)(tag & 7);
      if (num == 0) { ok = false; break; }
      if (num != cur) {
        // close the previous run: its last chunk keeps the bracket
        if (n > 0) s_tail[n - 1] = 0;
        // new run must be strictly ascending (non-adjacent duplicates are
        // the classic path's E_UNSUPPORTED -> host fallback)
        if (num < prev_num || n >= WG_DEC_MAX_ITEMS) { ok = false; break; }
        // field lookup decides chunkability (maps + per-entry repeated)
        cur_split = false;
        chunk_head = 0;
        for (int fi = 0; fi < tm.field_count; ++fi) {
          const FieldEntry& fe = t.fields[tm.field_start + fi];
          if (fe.number != num) continue;
          bool per_entry_len =
              (fe.flags & F_MAP) ||
              ((fe.flags & F_REPEATED) &&
               (fe.kind == K_MESSAGE || fe.kind == K_STRING ||
                fe.kind == K_BYTES));
          // unpacked repeated scalars (wt != W_LEN run) also chunk
          bool per_entry_scalar =
              (fe.flags & F_REPEATED) && !(fe.flags & F_MAP) && wt != W_LEN;
          if (per_entry_len || per_entry_scalar) {
            cur_split = true;
            // `"name":{` / `"name":[` = 1 + json_len + 2 + 1 bytes
            chunk_head = (uint32_t)fe.json_len + 4;
          }
          break;
        }
        s_head[n] = 0;
        s_tail[n] = (uint8_t)(cur_split ? 1 : 0);
        s_start[n++] = tag_start;
        prev_num = num;
        cur = num;
        chunk_wire = 0;
      } else if (cur_split && chunk_wire >= 4096) {
        // continuation chunk of the same run at an entry boundary
        if (n >= WG_DEC_MAX_ITEMS) { ok = false; break; }
        s_head[n] = (uint16_t)chunk_head;
        s_tail[n] = 1;
        s_start[n++] = tag_start;
        chunk_wire = 0;
      }
      if (wt == W_VARINT) {
        uint64_t v;
        if (!get_varint(pb, wire_len, &pos, &v)) { ok = false; break; }
      } else if (wt == W_I64) {
        if (pos + 8 > wire_len) { ok = false; break; }
        pos += 8;
      } else if (wt == W_I32) {
        if (pos + 4 > wire_len) { ok = false; break; }
        pos += 4;
      } else if (wt == W_LEN) {
        uint64_t v;
        if (!get_varint(pb, wire_len, &pos, &v)) { ok = false; break; }
        if (pos + v > wire_len) { ok = false; break; }
        pos += (uint32_t)v;
      } else {
        ok = false;
        break;
      }
      chunk_wire += pos - tag_start;
    }
    if (ok && pos == wire_len && n > 0) {
      s_tail[n - 1] = 0;  // the wire's last chunk keeps its bracket
      s_start[n] = wire_len;
      s_nitems = n;
    } else {
      s_mode = 0;  // classic single-wave fallback below
      s_nitems = 0;
    }
  }
  __syncthreads();
  const int n_items = s_nitems;

  // ---- phase B: decode items (dynamic wave grabs; the guard bound makes
  // the loop provably finite — a wave can win at most n_items grabs) ------
  if (s_mode && max_phase >= 1) {
    for (int guard = 0; guard <= WG_DEC_MAX_ITEMS + 1; ++guard) {
      int idx = 0;
      if (!lane) idx = atomicAdd(&s_next, 1);
      idx = __shfl(idx, 0, WAVE);
      if (idx >= n_items) break;
      uint32_t ist = s_start[idx], ien = s_start[idx + 1];
      uint32_t soff = 8u * ist + WG_DEC_ITEM_PAD * (uint32_t)idx;
      uint32_t scap = 8u * (ien - ist) + WG_DEC_ITEM_PAD;
      if (soff + scap > scr_cap) {
        if (!lane) {
          atomicCAS(&s_err, E_OK, E_OVERFLOW);
          s_outlen[idx] = 0;
        }
        continue;
      }
      DCtx c;
      c.pb = pb;
      c.len = ien;
      c.pos = ist;
      c.out = scr + soff;
      c.opos = 0;
      c.ocap = scap;
      c.t = t;
      c.status = E_OK;
      c.lane = lane;
      bool ok = decode_walk(c, msg_idx, ien, /*bare_top=*/true);
      uint32_t trim = (uint32_t)s_head[idx] + s_tail[idx];
      if (!ok || c.status != E_OK || c.opos < trim) {
        if (!lane) {
          atomicCAS(&s_err, E_OK, c.status == E_OK ? E_PARSE : c.status);
          s_outlen[idx] = 0;
        }
        continue;
      }
      // store the TRIMMED length (chunk joins drop re-emitted name syntax
      // and the premature closing bracket); src offsets add s_head later
      if (!lane) s_outlen[idx] = c.opos - trim;
    }
  }
  __syncthreads();

  // ---- phase C1: slice the item texts (one thread) ------------------------
  if (s_mode && threadIdx.x == 0) {
    s_next = 0;  // re-used by phase C2's grab loop
    int ns = 0;
    if (s_err == E_OK) {
      uint32_t sum = 0;
      for (int i = 0; i < n_items; ++i) sum += s_outlen[i];
      uint32_t sb = 4096;
      uint32_t budget = (uint32_t)(MAX_SLICES - n_items);
      if (budget > 0 && sum / budget + 1 > sb) sb = sum / budget + 1;
      s_slice_bytes = sb;
      bool clipped = false;
      for (int i = 0; i < n_items && !clipped; ++i) {
        for (uint32_t off = 0; off < s_outlen[i]; off += sb) {
          if (ns >= MAX_SLICES) {  // unreachable by the budget math above,
            clipped = true;        // but NEVER truncate output silently
            break;
          }
          s_sl_item[ns] = (uint16_t)i;
          s_sl_off[ns] = off;
          ++ns;
        }
      }
      if (clipped) s_err = E_OVERFLOW;
    }
    s_nslices = ns;
  }
  __syncthreads();
  const int n_slices = s_nslices;

  // ---- phase C2: escaped length per slice (wave-parallel) -----------------
  // static wave-strided assignment: slices are uniform-size, so dynamic
  // grabbing buys nothing over round-robin
  if (s_mode && s_err == E_OK && max_phase >= 2) {
    for (int s = wave; s < n_slices; s += WG_DEC_WAVES) {
      int item = s_sl_item[s];
      uint32_t ioff = s_sl_off[s];
      uint32_t ilen = s_outlen[item] - ioff;
      if (ilen > s_slice_bytes) ilen = s_slice_bytes;
      const uint8_t* src = scr + 8u * s_start[item] +
                           WG_DEC_ITEM_PAD * (uint32_t)item + s_head[item] +
                           ioff;
      uint32_t acc = 0;
      for (uint32_t i = lane; i < ilen; i += WAVE) acc += esc_len(src[i]);
      for (int d = WAVE / 2; d > 0; d >>= 1) acc += __shfl_down(acc, d, WAVE);
      if (!lane) s_sl_esc[s] = acc;
    }
  }
  __syncthreads();

  // ---- phase C3: prefix sums + envelope sizing (one thread) ---------------
  if (s_mode && threadIdx.x == 0) {
    s_next = 0;  // re-used by phase D's grab loop
    if (s_err == E_OK) {
      uint32_t idl = enc_results ? enc_results[req].id_len : 0;
      uint32_t pre = (uint32_t)(sizeof(WG_P1) - 1) + (idl ? idl : 4) +
                     (uint32_t)(sizeof(WG_P2) - 1) + 1 /* '{' */;
      uint32_t off = pre;
      int first = 1;
      for (int s = 0; s < n_slices; ++s) {
        uint8_t comma = 0;
        if (s_sl_off[s] == 0) {  // first slice of its item
          if (!first) comma = 1;
          first = 0;
        }
        s_sl_comma[s] = comma;
        off += comma;
        s_sl_fin[s] = off;
        off += s_sl_esc[s];
      }
      uint32_t total = off + 1 /* '}' */ + (uint32_t)(sizeof(WG_P3) - 1);
      if (total > fcap) s_err = E_OVERFLOW;
      s_total = total;
    }
  }
  __syncthreads();

  // ---- phase D: envelope framing (wave 0) + item escapes (all waves) ------
  if (s_mode && s_err == E_OK && max_phase >= 3) {
    if (wave == 0) {
      DCtx o;
      o.pb = nullptr;
      o.len = 0;
      o.pos = 0;
      o.out = fout;
      o.opos = 0;
      o.ocap = fcap;
      o.t = t;
      o.status = E_OK;
      o.lane = lane;
      bool ok = puts_(o, WG_P1, sizeof(WG_P1) - 1);
      uint32_t idl = enc_results ? enc_results[req].id_len : 0;
      if (ok && idl) {
        const uint8_t* idp = id_slots + (size_t)req * ID_SLOT_BYTES;
        for (uint32_t i = lane; i < idl; i += WAVE) o.out[o.opos + i] = idp[i];
        o.opos += idl;
      } else if (ok) {
        ok = puts_(o, "null", 4);
      }
      if (ok) ok = puts_(o, WG_P2, sizeof(WG_P2) - 1);
      if (ok) ok = putc_(o, '{');
      // closing '}' + P3 at the precomputed end
      uint32_t cpos = s_total - (uint32_t)(sizeof(WG_P3) - 1) - 1;
      if (!lane) fout[cpos] = '}';
      for (uint32_t i = lane; i < sizeof(WG_P3) - 1; i += WAVE)
        fout[cpos + 1 + i] = (uint8_t)WG_P3[i];
      if (!ok && !lane) atomicCAS(&s_err, E_OK, o.status);
    }
    for (int s = wave; s < n_slices; s += WG_DEC_WAVES) {
      int item = s_sl_item[s];
      uint32_t ioff = s_sl_off[s];
      uint32_t ilen = s_outlen[item] - ioff;
      if (ilen > s_slice_bytes) ilen = s_slice_bytes;
      const uint8_t* src = scr + 8u * s_start[item] +
                           WG_DEC_ITEM_PAD * (uint32_t)item + s_head[item] +
                           ioff;
      if (s_sl_comma[s] && !lane) fout[s_sl_fin[s] - 1] = ',';
      DCtx e;
      e.pb = nullptr;
      e.len = 0;
      e.pos = 0;
      e.out = fout + s_sl_fin[s];
      e.opos = 0;
      e.ocap = s_sl_esc[s];
      e.t = t;
      e.status = E_OK;
      e.lane = lane;
      // slice of already-validated walker output: skip the UTF-8 re-check
      // (a boundary can split a multi-byte char; escaping is per-byte)
      if (!put_escaped(e, src, ilen, /*validate=*/false)) {
        if (!lane) atomicCAS(&s_err, E_OK, e.status);
      }
    }
  } else if (!s_mode && wave == 0) {
    // ---- classic single-wave fallback (exact k_pb2json mode-0 body) ------
    DecodeResult r;
    r.status = E_OK;
    r.out_off = final_off[req];
    r.out_len = 0;
    r.pad = 0;
    DCtx c;
    c.pb = pb;
    c.len = wire_len;
    c.pos = 0;
    c.out = scr;
    c.opos = 0;
    c.ocap = scr_cap;
    c.t = t;
    c.status = E_OK;
    c.lane = lane;
    decode_walk(c, msg_idx, c.len);
    if (c.status != E_OK) {
      r.status = c.status;
      if (!lane) results[req] = r;
    } else {
      uint32_t json_len = c.opos;
      DCtx o;
      o.pb = nullptr;
      o.len = 0;
      o.pos = 0;
      o.out = fout;
      o.opos = 0;
      o.ocap = fcap;
      o.t = t;
      o.status = E_OK;
      o.lane = lane;
      bool ok = puts_(o, WG_P1, sizeof(WG_P1) - 1);
      uint32_t idl = enc_results ? enc_results[req].id_len : 0;
      if (ok && idl) {
        const uint8_t* idp = id_slots + (size_t)req * ID_SLOT_BYTES;
        for (uint32_t i = lane; i < idl; i += WAVE) o.out[o.opos + i] = idp[i];
        o.opos += idl;
      } else if (ok) {
        ok = puts_(o, "null", 4);
      }
      if (ok) ok = puts_(o, WG_P2, sizeof(WG_P2) - 1);
      if (ok) ok = put_escaped(o, scr, json_len);
      if (ok) ok = puts_(o, WG_P3, sizeof(WG_P3) - 1);
      if (!ok) {
        r.status = o.status;
      } else {
        r.status = E_OK;
        r.out_len = o.opos;
      }
      if (!lane) results[req] = r;
    }
  }
  __syncthreads();

  // ---- finalize (item mode) ------------------------------------------------
  if (s_mode && threadIdx.x == 0) {
    DecodeResult r;
    r.out_off = final_off[req];
    r.pad = 0;
    r.status = s_err;
    r.out_len = s_err == E_OK ? s_total : 0;
    results[req] = r;
  }
}
#endif  // GGRMCP_HOST_SIM
