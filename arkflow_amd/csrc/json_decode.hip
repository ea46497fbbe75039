// GPU JSON decode: one thread per document, extracting SCALAR fields of a
// KNOWN schema straight from the device-resident binary column into columnar
// outputs — the "GPU column builder kernel" of SURVEY §2.9 (the general
// schema-inference path stays host-side via pyarrow, per the §7 plan; this
// kernel covers the configured-schema hot path: flat JSON, numeric/bool
// fields, other keys skipped including nested objects/arrays/strings).
#include "common.h"

#define JSON_MAX_FIELDS 16
#define JSON_MAX_NAME 24

struct JsonSpec {
  int nf;
  char names[JSON_MAX_FIELDS][JSON_MAX_NAME];
  int name_len[JSON_MAX_FIELDS];
  int is_float[JSON_MAX_FIELDS];  // 1 → out_f64, 0 → out_i64 (ints + bools)
  int slot[JSON_MAX_FIELDS];
};

DEV_INLINE bool is_ws(uint8_t c) {
  return c == ' ' || c == '\t' || c == '\n' || c == '\r';
}

// skip a JSON value generically (string/number/object/array/literal)
DEV_INLINE int64_t skip_value(const uint8_t* d, int64_t p, int64_t end) {
  while (p < end && is_ws(d[p])) ++p;
  if (p >= end) return p;
  uint8_t c = d[p];
  if (c == '"') {
    ++p;
    while (p < end) {
      if (d[p] == '\\') p += 2;
      else if (d[p] == '"') return p + 1;
      else ++p;
    }
    return p;
  }
  if (c == '{' || c == '[') {
    uint8_t open = c, close = (c == '{') ? '}' : ']';
    int depth = 0;
    while (p < end) {
      uint8_t x = d[p];
      if (x == '"') {
        ++p;
        while (p < end) {
          if (d[p] == '\\') p += 2;
          else if (d[p] == '"') { ++p; break; }
          else ++p;
        }
        continue;
      }
      if (x == open) ++depth;
      else if (x == close) {
        --depth;
        if (depth == 0) return p + 1;
      }
      ++p;
    }
    return p;
  }
  // number / true / false / null
  while (p < end && d[p] != ',' && d[p] != '}' && d[p] != ']' &&
         !is_ws(d[p]))
    ++p;
  return p;
}

// parse a JSON number at p (after ws); returns value as double + intness
DEV_INLINE int64_t parse_number(const uint8_t* d, int64_t p, int64_t end,
                                double* out, bool* is_int) {
  bool neg = false;
  *is_int = true;
  if (p < end && (d[p] == '-' || d[p] == '+')) {
    neg = d[p] == '-';
    ++p;
  }
  double v = 0.0;
  while (p < end && d[p] >= '0' && d[p] <= '9') {
    v = v * 10.0 + (d[p] - '0');
    ++p;
  }
  if (p < end && d[p] == '.') {
    *is_int = false;
    ++p;
    double scale = 0.1;
    while (p < end && d[p] >= '0' && d[p] <= '9') {
      v += (d[p] - '0') * scale;
      scale *= 0.1;
      ++p;
    }
  }
  if (p < end && (d[p] == 'e' || d[p] == 'E')) {
    *is_int = false;
    ++p;
    bool eneg = false;
    if (p < end && (d[p] == '-' || d[p] == '+')) {
      eneg = d[p] == '-';
      ++p;
    }
    int ex = 0;
    while (p < end && d[p] >= '0' && d[p] <= '9') {
      ex = ex * 10 + (d[p] - '0');
      ++p;
    }
    double m = 1.0;
    for (int i = 0; i < ex; ++i) m *= 10.0;
    v = eneg ? v / m : v * m;
  }
  *out = neg ? -v : v;
  return p;
}

__global__ void json_decode_kernel(const uint8_t* __restrict__ data,
                                   const int64_t* __restrict__ offsets,
                                   int64_t n_docs, JsonSpec spec,
                                   double* __restrict__ out_f64,  // [nd][n]
                                   int64_t* __restrict__ out_i64,
                                   uint8_t* __restrict__ found,  // [nf][n]
                                   int32_t* __restrict__ err) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n_docs; i += stride) {
    int64_t p = offsets[i];
    const int64_t end = offsets[i + 1];
    while (p < end && is_ws(data[p])) ++p;
    if (p >= end || data[p] != '{') {
      if (p < end) err[0] = 1;
      continue;
    }
    ++p;
    for (;;) {
      while (p < end && (is_ws(data[p]) || data[p] == ',')) ++p;
      if (p >= end || data[p] == '}') break;
      if (data[p] != '"') { err[0] = 1; break; }
      // key
      int64_t k0 = ++p;
      while (p < end && data[p] != '"') {
        if (data[p] == '\\') ++p;
        ++p;
      }
      int klen = (int)(p - k0);
      ++p;  // closing quote
      while (p < end && is_ws(data[p])) ++p;
      if (p < end && data[p] == ':') ++p;
      while (p < end && is_ws(data[p])) ++p;
      // match against spec
      int fi = -1;
#pragma unroll
      for (int f = 0; f < JSON_MAX_FIELDS; ++f) {
        if (f < spec.nf && spec.name_len[f] == klen) {
          bool eq = true;
          for (int c = 0; c < klen; ++c)
            if (spec.names[f][c] != (char)data[k0 + c]) { eq = false; break; }
          if (eq) { fi = f; break; }
        }
      }
      if (fi < 0) {
        p = skip_value(data, p, end);
        continue;
      }
      uint8_t c = p < end ? data[p] : 0;
      double v = 0.0;
      bool is_int = true;
      if (c == 't') { v = 1.0; p += 4; }
      else if (c == 'f') { v = 0.0; p += 5; }
      else if (c == 'n') { p += 4; continue; }  // null → absent
      else if (c == '"') { p = skip_value(data, p, end); continue; }
      else p = parse_number(data, p, end, &v, &is_int);
      if (spec.is_float[fi])
        out_f64[(int64_t)spec.slot[fi] * n_docs + i] = v;
      else
        out_i64[(int64_t)spec.slot[fi] * n_docs + i] = (int64_t)v;
      found[(int64_t)fi * n_docs + i] = 1;
    }
  }
}

extern "C" void launch_json_decode(const uint8_t* data, const int64_t* offsets,
                                   int64_t n_docs, int nf, const char* names,
                                   const int* name_len, const int* is_float,
                                   const int* slot, double* out_f64,
                                   int64_t* out_i64, uint8_t* found,
                                   int32_t* err, hipStream_t st) {
  JsonSpec spec{};
  spec.nf = nf > JSON_MAX_FIELDS ? JSON_MAX_FIELDS : nf;
  for (int f = 0; f < spec.nf; ++f) {
    spec.name_len[f] = name_len[f];
    spec.is_float[f] = is_float[f];
    spec.slot[f] = slot[f];
    for (int c = 0; c < name_len[f] && c < JSON_MAX_NAME; ++c)
      spec.names[f][c] = names[f * JSON_MAX_NAME + c];
  }
  int grid = (int)((n_docs + 255) / 256);
  if (grid > 2048) grid = 2048;
  if (grid < 1) return;
  json_decode_kernel<<<grid, 256, 0, st>>>(data, offsets, n_docs, spec,
                                           out_f64, out_i64, found, err);
}
