// C++ fast predict path: PredictRequest bytes -> PredictResponse bytes
// with the GIL released for the whole call.
//
// The Python handler routes by model (ring/cache bookkeeping, metrics)
// and then hands the RAW request bytes here; this layer does the
// protobuf wire parse (the subset TF-Serving Predict uses:
// tensor_content-carrying TensorProtos), copies each input into pinned
// f32/i32 staging, DMAs to a device staging buffer, casts to the bf16
// workspace on-GPU, replays the ExecPlan hipGraph, casts/DMAs the
// outputs back and serializes the response — removing the per-request
// Python work that capped small-batch throughput.
//
// Anything it cannot handle (typed *_val fields, unknown dtypes,
// missing inputs) raises FastFallback and the Python path serves the
// request instead.
#include <torch/extension.h>

#include <hip/hip_runtime.h>

#include <cstring>
#include <atomic>
#include <chrono>
#include <condition_variable>
#include <map>
#include <memory>
#include <mutex>
#include <stdexcept>
#include <string>
#include <vector>

#include "kernels.h"
#include "fastpath_api.h"
#include "wire_parse.h"

namespace tfsc {

class ExecPlan;  // from executor.cpp
void fast_run_plan(void* plan, hipStream_t s);  // defined in executor.cpp

#define HIPCHK(x)                                                        \
  do {                                                                   \
    hipError_t e_ = (x);                                                 \
    if (e_ != hipSuccess)                                                \
      throw std::runtime_error(std::string("fastpath HIP error: ") +     \
                               hipGetErrorString(e_));                   \
  } while (0)

// ---------------------------------------------------------------------------
// minimal protobuf writer (PredictResponse)
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

static void w_len_prefixed(std::string& s, int fno,
                           const std::string& payload) {
  w_tag(s, fno, 2);
  w_varint(s, payload.size());
  s += payload;
}

// ---------------------------------------------------------------------------
// fast context/model
// ---------------------------------------------------------------------------
struct FastIO {
  std::string alias;
  bool is_int = false;            // i32 input (ids) vs f32<->bf16
  uintptr_t pin = 0;              // pinned host buffer (f32/i32)
  uintptr_t dev_stage = 0;        // device f32/i32 staging
  uintptr_t dev = 0;              // bf16 (or i32) workspace buffer
  int64_t row_elems = 0;          // elements per batch row
  std::vector<int64_t> tail_dims; // dims after the batch dim
};

struct FastContext {
  int bucket = 0;
  void* exec_plan = nullptr;      // tfsc::ExecPlan*
  hipStream_t stream = nullptr;
  std::vector<FastIO> ins;
  std::vector<FastIO> outs;
  std::mutex mu;
  bool disabled = false;
};

class FastModel {
 public:
  FastModel(std::string model_name, int64_t version, int target_ctxs)
      : name_(std::move(model_name)), version_(version),
        target_(target_ctxs < 1 ? 1 : target_ctxs) {}

  int add_context(int bucket, int target, uintptr_t exec_plan,
                  uintptr_t stream, std::vector<FastIO> ins,
                  std::vector<FastIO> outs) {
    auto ctx = std::make_unique<FastContext>();
    ctx->bucket = bucket;
    {
      std::lock_guard<std::mutex> g(mu_);
      if (target > 0) bucket_target_[bucket] = target;
    }
    ctx->exec_plan = reinterpret_cast<void*>(exec_plan);
    ctx->stream = reinterpret_cast<hipStream_t>(stream);
    ctx->ins = std::move(ins);
    ctx->outs = std::move(outs);
    std::lock_guard<std::mutex> g(mu_);
    if (specs_in_.empty()) {
      specs_in_ = ctx->ins;
      specs_out_ = ctx->outs;
    }
    ctxs_.push_back(std::move(ctx));
    return int(ctxs_.size()) - 1;
  }

  // server-side dynamic batching inside the fast path: requests whose
  // batch is below `merge_cap` wait up to `timeout_us` to merge with
  // concurrent requests into one plan execution (leader-follower)
  void enable_batching(int merge_cap, int64_t timeout_us) {
    merge_cap_ = merge_cap;
    timeout_us_ = timeout_us;
    batching_ = merge_cap > 1;
  }

  // shared-lock interface: the Python execution path locks the same
  // per-context mutex the fast path uses, so both paths serialize on
  // the context's stream/buffers
  void lock_ctx(int id) { ctxs_.at(id)->mu.lock(); }
  void unlock_ctx(int id) { ctxs_.at(id)->mu.unlock(); }

  // release path: waits out in-flight fast predicts, then blocks the
  // fast path permanently (buffers are about to be freed)
  void disable() {
    std::lock_guard<std::mutex> g(mu_);
    for (auto& c : ctxs_) {
      std::lock_guard<std::mutex> cg(c->mu);
      c->disabled = true;
    }
  }

  std::vector<FastIOSpec> input_specs() {
    std::lock_guard<std::mutex> g(mu_);
    std::vector<FastIOSpec> out;
    for (auto& io : specs_in_) {
      FastIOSpec sp;
      sp.alias = io.alias;
      sp.is_int = io.is_int;
      for (int64_t d : io.tail_dims) sp.tail.push_back(d);
      out.push_back(std::move(sp));
    }
    return out;
  }

  std::tuple<long long, long long, long long> stats() const {
    return {n_runs_.load(), n_requests_.load(), n_rows_.load()};
  }

  std::tuple<long long, long long, long long> stage_ns() const {
    return {ns_stage_in_.load(), ns_gpu_.load(), ns_serialize_.load()};
  }

  bool has_bucket(int batch) {
    std::lock_guard<std::mutex> g(mu_);
    for (auto& c : ctxs_)
      if (c->bucket >= batch) return true;
    return false;
  }

  struct Entry {
    const ParsedRequest* req;
    int64_t rows = 0;
    std::string out;
    int err = -1;                 // -1 pending, 0 ok, 1 fallback, 2 error
    std::string err_msg;
    bool promoted = false;        // follower promoted to leader
  };

  std::string predict(const uint8_t* data, size_t len) {
    ParsedRequest req = parse_request(data, len);
    int64_t rows = validate(req);
    Entry me;
    me.req = &req;
    me.rows = rows;
    if (!batching_ || rows >= merge_cap_) {
      Entry* one[1] = {&me};
      run_batch(one, 1);
    } else {
      grouped_predict(me);
    }
    if (me.err == 1) throw FastFallback(me.err_msg);
    if (me.err == 2) throw std::runtime_error(me.err_msg);
    return std::move(me.out);
  }

  int64_t validate(const ParsedRequest& req) {
    if (req.inputs.empty()) throw FastFallback("no inputs");
    int64_t batch = -1;
    for (auto& kv : req.inputs) {
      if (kv.second.has_typed_vals && !kv.second.content)
        throw FastFallback("typed-val tensor");
      if (!kv.second.content) throw FastFallback("no tensor_content");
      if (kv.second.dims.empty()) throw FastFallback("scalar input");
      if (batch < 0) batch = kv.second.dims[0];
      else if (batch != kv.second.dims[0])
        throw FastFallback("inconsistent batch");
    }
    if (batch < 0 || batch > (1 << 20))
      throw FastFallback("batch out of range");
    for (auto& kv : req.inputs)
      for (int64_t d : kv.second.dims)
        if (d < 0 || d > (1 << 24))
          throw FastFallback("dim out of range");
    std::lock_guard<std::mutex> g(mu_);
    if (specs_in_.empty()) throw FastFallback("no contexts yet");
    for (auto& io : specs_in_) {
      auto it = req.inputs.find(io.alias);
      if (it == req.inputs.end()) throw FastFallback("missing input");
      const ParsedTensor& t = it->second;
      if (int64_t(t.dims.size()) != int64_t(io.tail_dims.size()) + 1)
        throw FastFallback("rank mismatch");
      int64_t row = 1;
      for (size_t d = 0; d < io.tail_dims.size(); ++d) {
        if (t.dims[d + 1] != io.tail_dims[d])
          throw FastFallback("dim mismatch");
        row *= t.dims[d + 1];
      }
      if (io.is_int) {
        if (t.dtype != 3) throw FastFallback("want DT_INT32");
      } else if (t.dtype != 1) {
        throw FastFallback("want DT_FLOAT");
      }
      if (t.content_len != size_t(batch) * row * 4)
        throw FastFallback("content size");
    }
    return batch;
  }

  void grouped_predict(Entry& me) {
    // one merge group per output_filter signature
    std::string key;
    for (auto& f : me.req->output_filter) {
      key += f;
      key.push_back(0);
    }
    BatchGroup* g;
    {
      std::lock_guard<std::mutex> gm(groups_mu_);
      auto& slot = groups_[key];
      if (!slot) slot = std::make_unique<BatchGroup>();
      g = slot.get();
    }
    std::unique_lock<std::mutex> gl(g->m);
    g->q.push_back(&me);
    g->cv.notify_all();
    if (g->leader_active) {
      g->cv.wait(gl, [&] { return me.err != -1 || me.promoted; });
      if (me.err != -1) return;       // served (or failed) by a leader
      // promoted: fall through as the new leader
    } else {
      g->leader_active = true;
    }
    // leader: wait for followers until the window closes or full
    auto deadline = std::chrono::steady_clock::now() +
                    std::chrono::microseconds(timeout_us_);
    while (std::chrono::steady_clock::now() < deadline) {
      int64_t total = 0;
      for (auto* e : g->q) total += e->rows;
      if (total >= merge_cap_) break;
      if (g->cv.wait_until(gl, deadline) == std::cv_status::timeout)
        break;
    }
    // take a batch (cap by merge_cap_), leave the rest to a promoted
    // leader
    std::vector<Entry*> take;
    int64_t total = 0;
    size_t i = 0;
    for (; i < g->q.size(); ++i) {
      Entry* e = g->q[i];
      if (!take.empty() && total + e->rows > merge_cap_) break;
      take.push_back(e);
      total += e->rows;
    }
    g->q.erase(g->q.begin(), g->q.begin() + i);
    if (!g->q.empty()) {
      g->q.front()->promoted = true;
    } else {
      g->leader_active = false;
    }
    g->cv.notify_all();
    gl.unlock();

    run_batch(take.data(), take.size());

    gl.lock();
    g->cv.notify_all();
  }

  void run_batch(Entry** batch, size_t n) {
    try {
      run_batch_inner(batch, n);
      for (size_t i = 0; i < n; ++i)
        if (batch[i]->err == -1) batch[i]->err = 0;
    } catch (const FastFallback& e) {
      for (size_t i = 0; i < n; ++i) {
        batch[i]->err = 1;
        batch[i]->err_msg = e.what();
      }
    } catch (const std::exception& e) {
      for (size_t i = 0; i < n; ++i) {
        batch[i]->err = 2;
        batch[i]->err_msg = e.what();
      }
    }
  }

  void run_batch_inner(Entry** batch, size_t n) {
    int64_t total = 0;
    for (size_t i = 0; i < n; ++i) total += batch[i]->rows;
    n_runs_.fetch_add(1, std::memory_order_relaxed);
    n_requests_.fetch_add(static_cast<long long>(n), std::memory_order_relaxed);
    n_rows_.fetch_add(total, std::memory_order_relaxed);

    FastContext* ctx = acquire(int(total), n > 1);
    std::lock_guard<std::mutex> g2(ctx->mu, std::adopt_lock);
    if (ctx->disabled) throw FastFallback("model released");

    auto t0 = std::chrono::steady_clock::now();
    hipStream_t s = ctx->stream;
    for (auto& io : ctx->ins) {
      int64_t row = io.row_elems;
      size_t row_off = 0;
      for (size_t i = 0; i < n; ++i) {
        const ParsedTensor& t = batch[i]->req->inputs.at(io.alias);
        size_t nb = size_t(batch[i]->rows) * row * 4;
        std::memcpy(reinterpret_cast<char*>(io.pin) + row_off * row * 4,
                    t.content, nb);
        row_off += size_t(batch[i]->rows);
      }
      size_t want = size_t(total) * row * 4;
      if (io.is_int) {
        HIPCHK(hipMemcpyAsync(reinterpret_cast<void*>(io.dev),
                              reinterpret_cast<void*>(io.pin), want,
                              hipMemcpyHostToDevice, s));
        size_t cap = size_t(ctx->bucket) * row * 4;
        if (want < cap)
          HIPCHK(hipMemsetAsync(
              reinterpret_cast<char*>(io.dev) + want, 0, cap - want, s));
      } else {
        HIPCHK(hipMemcpyAsync(reinterpret_cast<void*>(io.dev_stage),
                              reinterpret_cast<void*>(io.pin), want,
                              hipMemcpyHostToDevice, s));
        launch_f32_to_bf16(s,
                           reinterpret_cast<const float*>(io.dev_stage),
                           reinterpret_cast<ushort*>(io.dev),
                           total * row);
        size_t cap2 = size_t(ctx->bucket) * row * 2;
        size_t used2 = size_t(total) * row * 2;
        if (used2 < cap2)
          HIPCHK(hipMemsetAsync(
              reinterpret_cast<char*>(io.dev) + used2, 0, cap2 - used2,
              s));
      }
    }

    auto t1 = std::chrono::steady_clock::now();
    fast_run_plan(ctx->exec_plan, s);

    const auto& filt = batch[0]->req->output_filter;
    std::vector<const FastIO*> wanted;
    for (auto& io : ctx->outs) {
      if (!filt.empty()) {
        bool keep = false;
        for (auto& f : filt)
          if (f == io.alias) keep = true;
        if (!keep) continue;
      }
      wanted.push_back(&io);
    }
    for (auto* io : wanted) {
      launch_bf16_to_f32(s, reinterpret_cast<const ushort*>(io->dev),
                         reinterpret_cast<float*>(io->dev_stage),
                         total * io->row_elems);
      HIPCHK(hipMemcpyAsync(reinterpret_cast<void*>(io->pin),
                            reinterpret_cast<void*>(io->dev_stage),
                            size_t(total) * io->row_elems * 4,
                            hipMemcpyDeviceToHost, s));
    }
    HIPCHK(hipStreamSynchronize(s));
    hipError_t ke = hipGetLastError();
    if (ke != hipSuccess)
      throw std::runtime_error(std::string("fastpath kernel error: ") +
                               hipGetErrorString(ke));
    auto t2 = std::chrono::steady_clock::now();
    using std::chrono::nanoseconds, std::chrono::duration_cast;
    ns_stage_in_.fetch_add(
        duration_cast<nanoseconds>(t1 - t0).count(),
        std::memory_order_relaxed);
    ns_gpu_.fetch_add(duration_cast<nanoseconds>(t2 - t1).count(),
                      std::memory_order_relaxed);

    // per-request responses from row slices of the pinned outputs
    size_t row_off = 0;
    for (size_t i = 0; i < n; ++i) {
      int64_t rows = batch[i]->rows;
      std::string& out = batch[i]->out;
      out.reserve(wanted.size() * 64 + 1024);
      for (auto* io : wanted) {
        std::string tp;
        w_tag(tp, 1, 0);
        w_varint(tp, 1);
        {
          std::string shape;
          {
            std::string dim;
            w_tag(dim, 1, 0);
            w_varint(dim, uint64_t(rows));
            w_len_prefixed(shape, 2, dim);
          }
          for (int64_t d : io->tail_dims) {
            std::string dim;
            w_tag(dim, 1, 0);
            w_varint(dim, uint64_t(d));
            w_len_prefixed(shape, 2, dim);
          }
          w_len_prefixed(tp, 2, shape);
        }
        size_t nbytes = size_t(rows) * io->row_elems * 4;
        w_tag(tp, 4, 2);
        w_varint(tp, nbytes);
        tp.append(reinterpret_cast<const char*>(io->pin) +
                      row_off * io->row_elems * 4,
                  nbytes);
        std::string entry;
        w_tag(entry, 1, 2);
        w_varint(entry, io->alias.size());
        entry += io->alias;
        w_len_prefixed(entry, 2, tp);
        w_len_prefixed(out, 1, entry);
      }
      {
        std::string spec;
        w_tag(spec, 1, 2);
        w_varint(spec, name_.size());
        spec += name_;
        std::string ver;
        w_tag(ver, 1, 0);
        w_varint(ver, uint64_t(version_));
        w_len_prefixed(spec, 2, ver);
        w_len_prefixed(out, 2, spec);
      }
      row_off += size_t(rows);
    }
    ns_serialize_.fetch_add(
        duration_cast<nanoseconds>(
            std::chrono::steady_clock::now() - t2).count(),
        std::memory_order_relaxed);
  }

 private:
  FastContext* acquire(int batch, bool block_if_warming = false) {
    // smallest registered bucket >= batch; prefer an idle context.
    // While fewer than `target_` contexts exist for that bucket and all
    // are busy, raise FastFallback so the Python path runs instead —
    // its contention handling builds (and then registers) the
    // remaining multi-stream contexts. At steady state, block
    // round-robin. Merged batches pass block_if_warming: a fallback
    // would explode the merge into per-request Python runs, and the
    // merge bucket is prewarmed anyway.
    int bucket = -1;
    std::vector<FastContext*> cand;
    {
      std::lock_guard<std::mutex> g(mu_);
      for (auto& c : ctxs_)
        if (c->bucket >= batch && (bucket < 0 || c->bucket < bucket))
          bucket = c->bucket;
      if (bucket < 0) throw FastFallback("no context for batch");
      for (auto& c : ctxs_) {
        if (c->bucket != bucket) continue;
        if (c->mu.try_lock()) return c.get();
        cand.push_back(c.get());
      }
    }
    int want = target_;
    {
      std::lock_guard<std::mutex> g(mu_);
      auto it = bucket_target_.find(bucket);
      if (it != bucket_target_.end()) want = it->second;
    }
    if (!block_if_warming && int(cand.size()) < want)
      throw FastFallback("contexts warming");
    FastContext* c = cand[rr_++ % cand.size()];
    c->mu.lock();
    return c;
  }

  struct BatchGroup {
    std::mutex m;
    std::condition_variable cv;
    std::vector<Entry*> q;
    bool leader_active = false;
  };

  std::string name_;
  int64_t version_;
  int target_;
  std::map<int, int> bucket_target_;
  std::atomic<unsigned> rr_{0};
  std::atomic<long long> n_runs_{0};       // plan executions
  std::atomic<long long> n_requests_{0};   // requests served
  std::atomic<long long> n_rows_{0};       // total rows executed
  // per-stage wall time (ns) — queue->H2D staging, GPU (H2D DMA +
  // kernels + D2H, stream-synchronized), response serialization
  std::atomic<long long> ns_stage_in_{0};
  std::atomic<long long> ns_gpu_{0};
  std::atomic<long long> ns_serialize_{0};
  std::mutex mu_;
  std::vector<std::unique_ptr<FastContext>> ctxs_;
  std::vector<FastIO> specs_in_, specs_out_;
  bool batching_ = false;
  int merge_cap_ = 1;
  int64_t timeout_us_ = 2000;
  std::mutex groups_mu_;
  std::map<std::string, std::unique_ptr<BatchGroup>> groups_;
};

// frontend.cpp entry points (fastpath_api.h)
std::string fastmodel_predict(FastModel* fm, const uint8_t* data,
                              size_t len) {
  return fm->predict(data, len);
}

std::vector<FastIOSpec> fastmodel_input_specs(FastModel* fm) {
  return fm->input_specs();
}


}  // namespace tfsc

namespace py = pybind11;

void register_fastpath(py::module_& mod) {
  using tfsc::FastIO;
  using tfsc::FastModel;

  py::register_exception<tfsc::FastFallback>(mod, "FastFallback");

  mod.def("peek_spec", [](py::bytes data) {
    char* buf = nullptr;
    Py_ssize_t blen = 0;
    if (PyBytes_AsStringAndSize(data.ptr(), &buf, &blen) != 0)
      throw py::error_already_set();
    std::string name, label;
    long long version = 0;
    bool has = tfsc::peek_spec_raw(
        reinterpret_cast<const uint8_t*>(buf), size_t(blen), &name,
        &version, &label);
    return py::make_tuple(py::str(name),
                          has ? py::cast(version) : py::none(),
                          py::str(label));
  });

  py::class_<FastIO>(mod, "FastIO")
      .def(py::init([](std::string alias, bool is_int, uintptr_t pin,
                       uintptr_t dev_stage, uintptr_t dev,
                       int64_t row_elems, std::vector<int64_t> tail) {
        FastIO io;
        io.alias = std::move(alias);
        io.is_int = is_int;
        io.pin = pin;
        io.dev_stage = dev_stage;
        io.dev = dev;
        io.row_elems = row_elems;
        io.tail_dims = std::move(tail);
        return io;
      }));

  py::class_<FastModel>(mod, "FastModel")
      .def(py::init<std::string, int64_t, int>())
      .def("add_context", &FastModel::add_context)
      .def("has_bucket", &FastModel::has_bucket)
      .def("stats", &FastModel::stats)
      .def("stage_ns", &FastModel::stage_ns)
      .def("_ptr", [](FastModel& fm) {
        return reinterpret_cast<uintptr_t>(&fm);
      })
      .def("enable_batching", &FastModel::enable_batching)
      .def("lock_ctx", &FastModel::lock_ctx,
           py::call_guard<py::gil_scoped_release>())
      .def("unlock_ctx", &FastModel::unlock_ctx)
      .def("disable", &FastModel::disable,
           py::call_guard<py::gil_scoped_release>())
      .def("predict", [](FastModel& fm, py::bytes data) {
        char* buf = nullptr;
        Py_ssize_t len = 0;
        if (PyBytes_AsStringAndSize(data.ptr(), &buf, &len) != 0)
          throw py::error_already_set();
        std::string out;
        {
          py::gil_scoped_release rel;
          out = fm.predict(reinterpret_cast<const uint8_t*>(buf),
                           size_t(len));
        }
        return py::bytes(out);
      });
}
