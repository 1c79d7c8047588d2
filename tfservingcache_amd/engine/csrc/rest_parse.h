// Dense-JSON parser for UNTRUSTED REST request bodies. Torch-free so
// the standalone ASan/UBSan fuzz harness (scripts/fuzz_parsers.cpp)
// can compile it; rest_frontend.cpp includes this for the serving
// path. NOTE: non-inline definitions — include from at most one TU
// per binary.
#pragma once

#include <cmath>
#include <cstdint>
#include <cstdlib>
#include <stdexcept>
#include <string>
#include <vector>

namespace tfsc_rest {

struct RestFallback : std::runtime_error {
  using std::runtime_error::runtime_error;
};

// ---------------------------------------------------------------------------
// tiny protobuf writer/reader (PredictRequest / PredictResponse subset)
// ---------------------------------------------------------------------------
static void w_varint(std::string& s, uint64_t v) {
  while (true) {
    uint8_t b = v & 0x7f;
    v >>= 7;
    if (v) s.push_back(char(b | 0x80));
    else { s.push_back(char(b)); return; }
  }
}
static void w_tag(std::string& s, int fno, int wt) {
  w_varint(s, uint64_t(fno) << 3 | wt);
}
static void w_len(std::string& s, int fno, const std::string& payload) {
  w_tag(s, fno, 2);
  w_varint(s, payload.size());
  s += payload;
}

struct PReader {
  const uint8_t* p;
  const uint8_t* end;
  uint64_t varint() {
    uint64_t v = 0;
    int shift = 0;
    while (p < end) {
      uint8_t b = *p++;
      v |= uint64_t(b & 0x7f) << shift;
      if (!(b & 0x80)) return v;
      shift += 7;
      if (shift >= 70) break;
    }
    throw std::runtime_error("bad varint in response");
  }
  uint64_t need(uint64_t n) {
    if (n > uint64_t(end - p)) throw std::runtime_error("truncated");
    return n;
  }
  void skip(int wt) {
    switch (wt) {
      case 0: varint(); break;
      case 1: p += 8; break;
      case 2: { uint64_t n = need(varint()); p += n; break; }
      case 5: p += 4; break;
      default: throw std::runtime_error("bad wire type");
    }
    if (p > end) throw std::runtime_error("truncated");
  }
};

struct OutTensor {
  std::string alias;
  std::vector<int64_t> dims;     // includes batch dim
  const uint8_t* data = nullptr; // f32 tensor_content
  size_t nbytes = 0;
};

// parse PredictResponse: outputs map (field 1) of TensorProto with
// dtype/shape/tensor_content
static std::vector<OutTensor> parse_response(const std::string& body) {
  std::vector<OutTensor> outs;
  PReader r{reinterpret_cast<const uint8_t*>(body.data()),
            reinterpret_cast<const uint8_t*>(body.data()) + body.size()};
  while (r.p < r.end) {
    uint64_t tag = r.varint();
    int fno = int(tag >> 3), wt = int(tag & 7);
    if (fno == 1 && wt == 2) {         // outputs entry
      uint64_t n = r.need(r.varint());
      PReader re{r.p, r.p + n};
      r.p += n;
      OutTensor t;
      while (re.p < re.end) {
        uint64_t etag = re.varint();
        if ((etag >> 3) == 1 && (etag & 7) == 2) {
          uint64_t kn = re.need(re.varint());
          t.alias.assign(reinterpret_cast<const char*>(re.p), kn);
          re.p += kn;
        } else if ((etag >> 3) == 2 && (etag & 7) == 2) {
          uint64_t vn = re.need(re.varint());
          PReader rt{re.p, re.p + vn};
          re.p += vn;
          while (rt.p < rt.end) {
            uint64_t ttag = rt.varint();
            int tf = int(ttag >> 3), tw = int(ttag & 7);
            if (tf == 2 && tw == 2) {            // tensor_shape
              uint64_t sn = rt.need(rt.varint());
              PReader rs{rt.p, rt.p + sn};
              rt.p += sn;
              while (rs.p < rs.end) {
                uint64_t stag = rs.varint();
                if ((stag >> 3) == 2 && (stag & 7) == 2) {
                  uint64_t dn = rs.need(rs.varint());
                  PReader rd{rs.p, rs.p + dn};
                  rs.p += dn;
                  int64_t size = 0;
                  while (rd.p < rd.end) {
                    uint64_t dtag = rd.varint();
                    if ((dtag >> 3) == 1 && (dtag & 7) == 0)
                      size = int64_t(rd.varint());
                    else rd.skip(int(dtag & 7));
                  }
                  t.dims.push_back(size);
                } else {
                  rs.skip(int(stag & 7));
                }
              }
            } else if (tf == 4 && tw == 2) {     // tensor_content
              uint64_t cn = rt.need(rt.varint());
              t.data = rt.p;
              t.nbytes = size_t(cn);
              rt.p += cn;
            } else {
              rt.skip(tw);
            }
          }
        } else {
          re.skip(int(etag & 7));
        }
      }
      outs.push_back(std::move(t));
    } else {
      r.skip(wt);
    }
  }
  return outs;
}

// ---------------------------------------------------------------------------
// dense-JSON parser: numeric nested arrays only; anything else throws
// RestFallback so the Python path serves the request
// ---------------------------------------------------------------------------
struct JParser {
  const char* p;
  const char* end;

  void ws() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' ||
                       *p == '\r'))
      ++p;
  }
  bool eat(char c) {
    ws();
    if (p < end && *p == c) { ++p; return true; }
    return false;
  }
  void expect(char c) {
    if (!eat(c)) throw RestFallback("json structure");
  }
  std::string string() {
    ws();
    if (p >= end || *p != '"') throw RestFallback("expected string");
    ++p;
    std::string out;
    while (p < end && *p != '"') {
      if (*p == '\\') {
        if (p + 1 >= end) throw RestFallback("bad escape");
        ++p;
        switch (*p) {
          case '"': out.push_back('"'); break;
          case '\\': out.push_back('\\'); break;
          case '/': out.push_back('/'); break;
          case 'n': out.push_back('\n'); break;
          case 't': out.push_back('\t'); break;
          case 'r': out.push_back('\r'); break;
          default: throw RestFallback("escape");   // \uXXXX etc -> python
        }
        ++p;
      } else {
        out.push_back(*p++);
      }
    }
    if (p >= end) throw RestFallback("unterminated string");
    ++p;
    return out;
  }
  // hand-rolled float parse (strtod is locale-aware and ~3-5x slower;
  // a 10 MB b=16 image body is mostly number text). Falls back to
  // strtod for exponents / >18-digit mantissas.
  double number() {
    ws();
    const char* start = p;
    bool neg = false;
    if (p < end && (*p == '-' || *p == '+')) {
      neg = (*p == '-');
      ++p;
    }
    uint64_t mant = 0;
    int digits = 0, frac = 0;
    while (p < end && *p >= '0' && *p <= '9') {
      mant = mant * 10 + uint64_t(*p - '0');
      ++digits;
      ++p;
    }
    if (p < end && *p == '.') {
      ++p;
      while (p < end && *p >= '0' && *p <= '9') {
        mant = mant * 10 + uint64_t(*p - '0');
        ++digits;
        ++frac;
        ++p;
      }
    }
    if (digits == 0) throw RestFallback("expected number");
    if (digits > 18 || (p < end && (*p == 'e' || *p == 'E'))) {
      char* np = nullptr;
      double v = strtod(start, &np);
      if (np == start) throw RestFallback("expected number");
      p = np;
      return v;
    }
    static const double kPow10[19] = {
        1e0, 1e1, 1e2, 1e3, 1e4, 1e5, 1e6, 1e7, 1e8, 1e9, 1e10, 1e11,
        1e12, 1e13, 1e14, 1e15, 1e16, 1e17, 1e18};
    double v = double(mant) / kPow10[frac];
    return neg ? -v : v;
  }
};

// parse a nested numeric array; dims[d] is the (uniform) length of the
// arrays at depth d. Inner arrays complete before outer ones, so dims
// slots are filled innermost-first: grow with -1 sentinels and assign
// on first completion at each depth, enforce equality afterwards.
// Rectangularity of the element COUNT vs the dims product is
// re-checked downstream (FastModel::validate), so rare shapes that
// slip through here still fall back cleanly.
static void parse_dense(JParser& j, int depth,
                        std::vector<int64_t>& dims,
                        std::vector<float>& fdata,
                        std::vector<int32_t>& idata, bool is_int) {
  // recursion depth == tensor rank; cap it or a body of 100k '['s
  // overflows the native stack (remote crash). Real tensors are <= 8-D.
  if (depth > 32) throw RestFallback("nesting too deep");
  j.ws();
  if (j.p < j.end && *j.p == '[') {
    ++j.p;
    int64_t count = 0;
    j.ws();
    if (j.p < j.end && *j.p == ']') { ++j.p; }
    else if (j.p < j.end && *j.p != '[') {
      // innermost array: tight inline number loop (the bulk of a
      // multi-MB image body) — no recursion/function calls per leaf
      static const double kPow10[19] = {
          1e0, 1e1, 1e2, 1e3, 1e4, 1e5, 1e6, 1e7, 1e8, 1e9, 1e10,
          1e11, 1e12, 1e13, 1e14, 1e15, 1e16, 1e17, 1e18};
      const char* p = j.p;
      const char* end = j.end;
      while (true) {
        const char* start = p;
        bool neg = false;
        if (p < end && (*p == '-' || *p == '+')) {
          neg = (*p == '-');
          ++p;
        }
        uint64_t mant = 0;
        int digits = 0, frac = 0;
        while (p < end && unsigned(*p - '0') < 10u) {
          mant = mant * 10 + uint64_t(*p - '0');
          ++digits;
          ++p;
        }
        if (p < end && *p == '.') {
          ++p;
          while (p < end && unsigned(*p - '0') < 10u) {
            mant = mant * 10 + uint64_t(*p - '0');
            ++digits;
            ++frac;
            ++p;
          }
        }
        double v;
        if (digits == 0 || digits > 18 ||
            (p < end && (*p == 'e' || *p == 'E'))) {
          char* np = nullptr;
          v = strtod(start, &np);
          if (np == start) throw RestFallback("expected number");
          p = np;
        } else {
          v = double(mant) / kPow10[frac];
          if (neg) v = -v;
        }
        if (is_int) idata.push_back(int32_t(llround(v)));
        else fdata.push_back(float(v));
        ++count;
        while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' ||
                           *p == '\r'))
          ++p;
        if (p < end && *p == ',') {
          ++p;
          while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' ||
                             *p == '\r'))
            ++p;
          continue;
        }
        break;
      }
      if (p >= end || *p != ']') throw RestFallback("expected ]");
      ++p;
      j.p = p;
      // leaves live one level deeper than this array
      if (int(dims.size()) > depth + 1)
        throw RestFallback("ragged depth");
    }
    else {
      while (true) {
        parse_dense(j, depth + 1, dims, fdata, idata, is_int);
        ++count;
        j.ws();
        if (j.p < j.end && *j.p == ',') { ++j.p; continue; }
        break;
      }
      if (!j.eat(']')) throw RestFallback("expected ]");
    }
    if (int(dims.size()) <= depth) dims.resize(depth + 1, -1);
    if (dims[depth] == -1) dims[depth] = count;
    else if (dims[depth] != count) throw RestFallback("ragged");
    return;
  }
  double v = j.number();
  // a number at depth d is ragged if any ARRAY exists at depth >= d
  if (int(dims.size()) > depth) throw RestFallback("ragged depth");
  if (is_int) idata.push_back(int32_t(llround(v)));
  else fdata.push_back(float(v));
}

static void check_dims_complete(const std::vector<int64_t>& dims) {
  for (int64_t d : dims)
    if (d < 0) throw RestFallback("incomplete dims");
}


}  // namespace tfsc_rest
