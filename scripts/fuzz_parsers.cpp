// ASan/UBSan fuzz harness for the UNTRUSTED-input parsers (SURVEY §5
// race-detection/sanitizer gap): the protobuf wire reader the C++ fast
// predict path runs on raw request bytes (wire_parse.h) and the
// dense-JSON parser of the native REST front-end (rest_parse.h).
//
// Build + run (no GPU, no torch — tests/test_sanitizers.py does this):
//   g++ -std=c++17 -O1 -g -fsanitize=address,undefined \
//       -fno-sanitize-recover=all \
//       -I tfservingcache_amd/engine/csrc \
//       scripts/fuzz_parsers.cpp -o /tmp/fuzz_parsers
//   /tmp/fuzz_parsers [iterations]
//
// Strategy: structured seeds (valid PredictRequests / JSON bodies)
// mutated by random truncation, byte flips and splices, plus pure
// random bytes. Any OOB read/write, overflow or UB aborts the binary.
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <random>
#include <string>
#include <vector>

#include "wire_parse.h"
#include "rest_parse.h"

namespace {

std::mt19937_64 rng(0xC0FFEE);

// -- seed builders ----------------------------------------------------------
void w_varint(std::string& s, uint64_t v) {
  while (true) {
    uint8_t b = v & 0x7f;
    v >>= 7;
    if (v) s.push_back(char(b | 0x80));
    else { s.push_back(char(b)); return; }
  }
}
void w_tag(std::string& s, int fno, int wt) {
  w_varint(s, uint64_t(fno) << 3 | wt);
}
void w_len(std::string& s, int fno, const std::string& p) {
  w_tag(s, fno, 2);
  w_varint(s, p.size());
  s += p;
}

std::string seed_predict_request(int rows, int cols) {
  std::string tp;
  w_tag(tp, 1, 0);
  w_varint(tp, 1);                      // DT_FLOAT
  std::string shape;
  for (int d : {rows, cols}) {
    std::string dim;
    w_tag(dim, 1, 0);
    w_varint(dim, uint64_t(d));
    w_len(shape, 2, dim);
  }
  w_len(tp, 2, shape);
  std::vector<float> vals(size_t(rows) * cols, 0.5f);
  w_tag(tp, 4, 2);
  w_varint(tp, vals.size() * 4);
  tp.append(reinterpret_cast<const char*>(vals.data()), vals.size() * 4);

  std::string spec;
  w_tag(spec, 1, 2);
  w_varint(spec, 5);
  spec += "model";
  std::string ver;
  w_tag(ver, 1, 0);
  w_varint(ver, 3);
  w_len(spec, 2, ver);

  std::string entry;
  w_tag(entry, 1, 2);
  w_varint(entry, 1);
  entry += "x";
  w_len(entry, 2, tp);

  std::string req;
  w_len(req, 1, spec);
  w_len(req, 2, entry);
  w_tag(req, 3, 2);
  w_varint(req, 1);
  req += "y";
  return req;
}

std::string seed_json(int rows, int cols) {
  std::string s = "[";
  for (int r = 0; r < rows; ++r) {
    if (r) s += ",";
    s += "[";
    for (int c = 0; c < cols; ++c) {
      if (c) s += ",";
      s += "-0.123456";
    }
    s += "]";
  }
  s += "]";
  return s;
}

std::string mutate(const std::string& seed) {
  std::string s = seed;
  switch (rng() % 5) {
    case 0:                              // truncate
      s.resize(rng() % (s.size() + 1));
      break;
    case 1: {                            // flip bytes
      for (int i = 0; i < 8 && !s.empty(); ++i)
        s[rng() % s.size()] = char(rng());
      break;
    }
    case 2: {                            // splice a chunk elsewhere
      if (s.size() > 8) {
        size_t a = rng() % s.size(), b = rng() % s.size();
        size_t n = 1 + rng() % 16;
        for (size_t i = 0; i < n && a + i < s.size() && b + i < s.size();
             ++i)
          s[a + i] = s[b + i];
      }
      break;
    }
    case 3: {                            // pure random bytes
      s.resize(rng() % 300);
      for (auto& ch : s) ch = char(rng());
      break;
    }
    case 4: {                            // inflate a random varint-ish
      if (!s.empty()) s[rng() % s.size()] = char(0xFF);
      break;
    }
  }
  return s;
}

void drive_wire(const std::string& data) {
  const uint8_t* p = reinterpret_cast<const uint8_t*>(data.data());
  try {
    std::string name, label;
    long long ver = 0;
    tfsc::peek_spec_raw(p, data.size(), &name, &ver, &label);
  } catch (const tfsc::FastFallback&) {
  }
  try {
    auto req = tfsc::parse_request(p, data.size());
    // touch what the fast path would: dims + content of every input
    for (auto& kv : req.inputs) {
      volatile int64_t acc = 0;
      for (int64_t d : kv.second.dims) acc += d;
      if (kv.second.content && kv.second.content_len) {
        // content must lie inside the buffer (the round-1 OOB class)
        const uint8_t* c = kv.second.content;
        if (c < p || c + kv.second.content_len > p + data.size()) {
          fprintf(stderr, "content escapes buffer!\n");
          abort();
        }
        volatile uint8_t x = c[kv.second.content_len - 1];
        (void)x;
      }
      (void)acc;
    }
  } catch (const tfsc::FastFallback&) {
  }
}

void drive_json(const std::string& data) {
  try {
    tfsc_rest::JParser j{data.data(), data.data() + data.size()};
    std::vector<int64_t> dims;
    std::vector<float> f;
    std::vector<int32_t> iv;
    tfsc_rest::parse_dense(j, 0, dims, f, iv, (rng() & 1) != 0);
    tfsc_rest::check_dims_complete(dims);
  } catch (const tfsc_rest::RestFallback&) {
  }
}

}  // namespace

int main(int argc, char** argv) {
  long iters = argc > 1 ? atol(argv[1]) : 20000;
  std::vector<std::string> wire_seeds = {
      seed_predict_request(2, 8), seed_predict_request(1, 1),
      seed_predict_request(4, 64)};
  std::vector<std::string> json_seeds = {
      seed_json(2, 8), "[1,2,3]", "[[1],[2]]",
      "[1e300, -0.5, 448, 0.00001]", "[[\"x\"]]", "[[[[1]]]]",
      // deep-nesting class (depth cap regression: a crafted body must
      // throw RestFallback, never overflow the native stack)
      std::string(1000, '[') + "1" + std::string(1000, ']'),
      std::string(64, '[') + "1,2" + std::string(64, ']')};
  // directed deep-nesting sweep beyond the mutation loop
  for (int d : {16, 33, 100, 5000, 200000}) {
    drive_json(std::string(size_t(d), '[') + "1" +
               std::string(size_t(d), ']'));
    drive_json(std::string(size_t(d), '['));
  }
  for (long i = 0; i < iters; ++i) {
    drive_wire(mutate(wire_seeds[rng() % wire_seeds.size()]));
    drive_json(mutate(json_seeds[rng() % json_seeds.size()]));
  }
  printf("fuzz_parsers: %ld iterations clean\n", iters);
  return 0;
}
