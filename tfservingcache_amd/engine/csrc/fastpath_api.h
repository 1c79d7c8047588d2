// Shared surface between the C++ fast predict path (fastpath.cpp) and
// the native gRPC front-end (frontend.cpp): the fallback exception, an
// opaque FastModel handle, and the raw model-spec peek.
#pragma once

#include <cstddef>
#include <cstdint>
#include <stdexcept>
#include <string>
#include <vector>

namespace tfsc {

struct FastFallback : std::runtime_error {
  using std::runtime_error::runtime_error;
};

class FastModel;  // full definition in fastpath.cpp

// input-tensor schema of a registered model (for the native REST
// front-end's JSON -> proto bridge)
struct FastIOSpec {
  std::string alias;
  bool is_int = false;
  std::vector<long long> tail;     // dims after the batch dim
};
std::vector<FastIOSpec> fastmodel_input_specs(FastModel* fm);

// PredictRequest bytes -> PredictResponse bytes entirely in C++
// (throws FastFallback when the request needs the Python path).
std::string fastmodel_predict(FastModel* fm, const uint8_t* data,
                              size_t len);

// Partial parse of PredictRequest.model_spec (field 1). Returns false
// if no version was present (serve-latest semantics).
bool peek_spec_raw(const uint8_t* p, size_t len, std::string* name,
                   long long* version, std::string* label);

}  // namespace tfsc
