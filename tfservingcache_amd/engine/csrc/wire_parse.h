// Wire-format parsers for UNTRUSTED request bytes (PredictRequest
// subset + model_spec peek). Extracted into a torch-free header so a
// standalone ASan/UBSan fuzz harness (scripts/fuzz_parsers.cpp,
// exercised by tests/test_sanitizers.py) can compile them without the
// HIP/torch toolchain. fastpath.cpp includes this for the serving path.
#pragma once

#include <cstdint>
#include <cstring>
#include <map>
#include <string>
#include <vector>

#include "fastpath_api.h"

namespace tfsc {

// ---------------------------------------------------------------------------
// minimal protobuf wire reader
// ---------------------------------------------------------------------------
struct Reader {
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
    throw FastFallback("bad varint");
  }

  // bounds check before consuming n length-delimited bytes: a
  // truncated-but-internally-consistent request must throw here, never
  // read past the buffer (remote OOB read otherwise)
  uint64_t need(uint64_t n) {
    if (n > uint64_t(end - p)) throw FastFallback("truncated field");
    return n;
  }

  void skip(int wt) {
    switch (wt) {
      case 0: varint(); break;
      case 1: p += 8; break;
      case 2: { uint64_t n = need(varint()); p += n; break; }
      case 5: p += 4; break;
      default: throw FastFallback("bad wire type");
    }
    if (p > end) throw FastFallback("truncated");
  }
};

struct ParsedTensor {
  int dtype = 0;
  std::vector<int64_t> dims;
  const uint8_t* content = nullptr;
  size_t content_len = 0;
  bool has_typed_vals = false;
};

static ParsedTensor parse_tensor(const uint8_t* p, const uint8_t* end) {
  ParsedTensor t;
  Reader r{p, end};
  while (r.p < r.end) {
    uint64_t tag = r.varint();
    int fno = int(tag >> 3), wt = int(tag & 7);
    if (fno == 1 && wt == 0) {
      t.dtype = int(r.varint());
    } else if (fno == 2 && wt == 2) {            // tensor_shape
      uint64_t n = r.need(r.varint());
      Reader rs{r.p, r.p + n};
      r.p += n;
      while (rs.p < rs.end) {
        uint64_t stag = rs.varint();
        if ((stag >> 3) == 2 && (stag & 7) == 2) {   // dim
          uint64_t dn = rs.need(rs.varint());
          Reader rd{rs.p, rs.p + dn};
          rs.p += dn;
          int64_t size = 0;
          while (rd.p < rd.end) {
            uint64_t dtag = rd.varint();
            if ((dtag >> 3) == 1 && (dtag & 7) == 0)
              size = int64_t(rd.varint());
            else
              rd.skip(int(dtag & 7));
          }
          t.dims.push_back(size);
        } else {
          rs.skip(int(stag & 7));
        }
      }
    } else if (fno == 4 && wt == 2) {            // tensor_content
      uint64_t n = r.need(r.varint());
      t.content = r.p;
      t.content_len = size_t(n);
      r.p += n;
    } else if (fno >= 5 && fno <= 17) {
      t.has_typed_vals = true;
      r.skip(wt);
    } else {
      r.skip(wt);
    }
  }
  return t;
}

struct ParsedRequest {
  std::map<std::string, ParsedTensor> inputs;
  std::vector<std::string> output_filter;
};

static ParsedRequest parse_request(const uint8_t* p, size_t len) {
  ParsedRequest req;
  Reader r{p, p + len};
  while (r.p < r.end) {
    uint64_t tag = r.varint();
    int fno = int(tag >> 3), wt = int(tag & 7);
    if (fno == 2 && wt == 2) {        // inputs map entry
      uint64_t n = r.need(r.varint());
      Reader re{r.p, r.p + n};
      r.p += n;
      std::string key;
      const uint8_t* vptr = nullptr;
      size_t vlen = 0;
      while (re.p < re.end) {
        uint64_t etag = re.varint();
        if ((etag >> 3) == 1 && (etag & 7) == 2) {
          uint64_t kn = re.need(re.varint());
          key.assign(reinterpret_cast<const char*>(re.p), kn);
          re.p += kn;
        } else if ((etag >> 3) == 2 && (etag & 7) == 2) {
          uint64_t vn = re.need(re.varint());
          vptr = re.p;
          vlen = size_t(vn);
          re.p += vn;
        } else {
          re.skip(int(etag & 7));
        }
      }
      if (vptr) req.inputs[key] = parse_tensor(vptr, vptr + vlen);
    } else if (fno == 3 && wt == 2) {  // output_filter
      uint64_t n = r.need(r.varint());
      req.output_filter.emplace_back(
          reinterpret_cast<const char*>(r.p), n);
      r.p += n;
    } else {
      r.skip(wt);
    }
  }
  return req;
}

// fast partial parse of PredictRequest.model_spec (field 1): the
// Python routing layer (and the native front-end's registry lookup)
// use this instead of a full decode
bool peek_spec_raw(const uint8_t* p, size_t len, std::string* name,
                   long long* version, std::string* label) {
  int64_t ver = 0;
  bool has_version = false;
  Reader r{p, p + len};
  while (r.p < r.end) {
    uint64_t tag = r.varint();
    if ((tag >> 3) == 1 && (tag & 7) == 2) {
      uint64_t n = r.need(r.varint());
      Reader rs{r.p, r.p + n};
      r.p += n;
      while (rs.p < rs.end) {
        uint64_t stag = rs.varint();
        int fno = int(stag >> 3), wt = int(stag & 7);
        if (fno == 1 && wt == 2) {
          uint64_t kn = rs.need(rs.varint());
          name->assign(reinterpret_cast<const char*>(rs.p), kn);
          rs.p += kn;
        } else if (fno == 2 && wt == 2) {          // Int64Value version
          uint64_t vn = rs.need(rs.varint());
          Reader rv{rs.p, rs.p + vn};
          rs.p += vn;
          while (rv.p < rv.end) {
            uint64_t vtag = rv.varint();
            if ((vtag >> 3) == 1 && (vtag & 7) == 0) {
              ver = int64_t(rv.varint());
              has_version = true;
            } else {
              rv.skip(int(vtag & 7));
            }
          }
        } else if (fno == 4 && wt == 2) {
          uint64_t kn = rs.need(rs.varint());
          label->assign(reinterpret_cast<const char*>(rs.p), kn);
          rs.p += kn;
        } else {
          rs.skip(wt);
        }
      }
      break;
    }
    r.skip(int(tag & 7));
  }
  *version = ver;
  return has_version;
}

}  // namespace tfsc
