// ExecPlan: a compiled sequence of kernel launches for one (model, batch
// bucket). The Python planner (engine/gpu.py) resolves every shape,
// buffer offset and fusion decision ahead of time; this layer only
// launches CDNA4 kernels in order on the current HIP stream, so the
// per-request overhead is one Python->C++ call. hipGraph capture/replay
// of the whole sequence removes the per-kernel launch overhead for
// launch-bound small models (MI355X_MICROARCH.md §price list: boundary).
//
// This (plus the .hip kernels) replaces the graph execution the
// reference delegated to tensorflow_model_server.
#include <torch/extension.h>

#include <hip/hip_runtime.h>

#include <c10/hip/HIPStream.h>

#include <cstdint>
#include <stdexcept>
#include <vector>

#include "kernels.h"

namespace tfsc {

enum CallKind : int {
  K_ELT_UNARY = 0,
  K_ELT_BINARY,
  K_BN_ACT,
  K_SOFTMAX,
  K_LAYERNORM,
  K_MEAN_LAST,
  K_MEAN_MID,
  K_POOL,
  K_TRANSPOSE,
  K_GATHER,
  K_PAD_NHWC,
  K_GEMM,
  K_BGEMM,
  K_IM2COL,
  K_CONV,
  K_PAD_LAST,
  K_ATTENTION,
  K_DEPTHWISE,
  K_GEMM_FP8,
  K_QUANT_FP8,
  K_CAST,
  K_ARGMAX_LAST,
  K_SCATTER,
};

struct Call {
  int kind;
  std::vector<intptr_t> ptrs;
  std::vector<int64_t> ints;
  std::vector<float> floats;
};

static void launch_call(const Call& c, hipStream_t s) {
  auto p = [&](int i) { return reinterpret_cast<ushort*>(c.ptrs[i]); };
  auto cp = [&](int i) { return reinterpret_cast<const ushort*>(c.ptrs[i]); };
  const auto& I = c.ints;
  switch (c.kind) {
    case K_ELT_UNARY:
      launch_eltwise_unary(s, cp(0), p(1), I[0], int(I[1]));
      break;
    case K_ELT_BINARY: {
      // ints layout: [n_out, fn, ndim, dims[ndim], sa[ndim], sb[ndim]]
      BcastArgs bc;
      bc.ndim = int(I[2]);
      for (int d = 0; d < MAX_DIMS; ++d) {
        bc.dims[d] = d < bc.ndim ? I[3 + d] : 1;
        bc.sa[d] = d < bc.ndim ? I[3 + bc.ndim + d] : 0;
        bc.sb[d] = d < bc.ndim ? I[3 + 2 * bc.ndim + d] : 0;
      }
      launch_eltwise_binary(s, cp(0), cp(1), p(2), I[0], bc, int(I[1]));
      break;
    }
    case K_BN_ACT:
      launch_bn_act(s, cp(0), cp(1), cp(2), p(3), I[0], I[1], int(I[2]));
      break;
    case K_SOFTMAX:
      launch_softmax(s, cp(0), p(1), I[0], I[1]);
      break;
    case K_LAYERNORM:
      launch_layernorm(s, cp(0), cp(1), cp(2), p(3), I[0], I[1],
                       c.floats[0]);
      break;
    case K_MEAN_LAST:
      launch_reduce_mean_last(s, cp(0), p(1), I[0], I[1]);
      break;
    case K_MEAN_MID:
      launch_reduce_mean_mid(s, cp(0), p(1), I[0], I[1], I[2]);
      break;
    case K_POOL:
      launch_pool(s, cp(0), p(1), I[0] != 0, int(I[1]), int(I[2]),
                  int(I[3]), int(I[4]), int(I[5]), int(I[6]), int(I[7]),
                  int(I[8]), int(I[9]), int(I[10]), int(I[11]), int(I[12]));
      break;
    case K_TRANSPOSE: {
      int ndim = int(I[0]);
      launch_transpose(s, cp(0), p(1), ndim, I.data() + 1,
                       I.data() + 1 + ndim, I[1 + 2 * ndim]);
      break;
    }
    case K_GATHER:
      launch_gather_rows(s, cp(0), reinterpret_cast<const int*>(c.ptrs[1]),
                         p(2), I[0], I[1]);
      break;
    case K_PAD_NHWC:
      launch_pad_nhwc(s, cp(0), p(1), int(I[0]), int(I[1]), int(I[2]),
                      int(I[3]), int(I[4]), int(I[5]), int(I[6]),
                      int(I[7]));
      break;
    case K_GEMM: {
      const ushort* bias = c.ptrs[2] ? cp(2) : nullptr;
      const ushort* res = c.ptrs[3] ? cp(3) : nullptr;
      launch_gemm(s, cp(0), cp(1), bias, res, p(4), I[0], I[1], I[2],
                  int(I[3]), c.floats[0]);
      break;
    }
    case K_BGEMM:
      launch_batched_gemm(s, cp(0), cp(1), p(2), I[0], I[1], I[2], I[3],
                          I[4], I[5], I[6], I[7] != 0, c.floats[0]);
      break;
    case K_IM2COL:
      launch_im2col(s, cp(0), p(1), int(I[0]), int(I[1]), int(I[2]),
                    int(I[3]), int(I[4]), int(I[5]), int(I[6]), int(I[7]),
                    int(I[8]), int(I[9]), int(I[10]), int(I[11]),
                    int(I[12]));
      break;
    case K_CONV: {
      // ptrs: [x, w, bias, residual|0, zeros, y]
      // ints: [N,H,W,C,Kc,R,S,sh,sw,pt,pl,Ho,Wo,k_pad,act]
      const ushort* res = c.ptrs[3] ? cp(3) : nullptr;
      launch_conv_igemm(s, cp(0), cp(1), cp(2), res, cp(4), p(5),
                        int(I[0]), int(I[1]), int(I[2]), int(I[3]),
                        int(I[4]), int(I[5]), int(I[6]), int(I[7]),
                        int(I[8]), int(I[9]), int(I[10]), int(I[11]),
                        int(I[12]), int(I[13]), int(I[14]));
      break;
    }
    case K_PAD_LAST:
      launch_pad_last(s, cp(0), p(1), I[0], int(I[1]), int(I[2]));
      break;
    case K_ATTENTION:
      // ptrs: [q, k, v, out]; ints: [B, S, H, D]; floats: [scale]
      launch_attention(s, cp(0), cp(1), cp(2), p(3), int(I[0]), int(I[1]),
                       int(I[2]), int(I[3]), c.floats[0]);
      break;
    case K_GEMM_FP8: {
      // ptrs: [Aq, sa, Bq, sb, bias|0, residual|0, C]
      // ints: [M, N, Kp, act]
      const ushort* bias8 = c.ptrs[4] ? cp(4) : nullptr;
      const ushort* res8 = c.ptrs[5] ? cp(5) : nullptr;
      launch_gemm_fp8(s, reinterpret_cast<const uint8_t*>(c.ptrs[0]),
                      reinterpret_cast<const float*>(c.ptrs[1]),
                      reinterpret_cast<const uint8_t*>(c.ptrs[2]),
                      reinterpret_cast<const float*>(c.ptrs[3]),
                      bias8, res8, p(6), I[0], I[1], I[2], int(I[3]));
      break;
    }
    case K_QUANT_FP8:
      // ptrs: [x(bf16), q(u8), scales(f32)]; ints: [M, K, Kp]
      launch_quant_rowwise(s, cp(0),
                           reinterpret_cast<uint8_t*>(c.ptrs[1]),
                           reinterpret_cast<float*>(c.ptrs[2]),
                           I[0], I[1], I[2]);
      break;
    case K_SCATTER: {
      // ints: [ndim, in_dims..., out_strides..., n_in]
      int nd = int(I[0]);
      launch_scatter(s, cp(0), p(1), nd, I.data() + 1, I.data() + 1 + nd,
                     I[1 + 2 * nd]);
      break;
    }
    case K_CAST:
      launch_cast(s, reinterpret_cast<const void*>(c.ptrs[0]),
                  reinterpret_cast<void*>(c.ptrs[1]), I[0], int(I[1]));
      break;
    case K_ARGMAX_LAST:
      launch_argmax_last(s, cp(0), reinterpret_cast<int*>(c.ptrs[1]),
                         I[0], I[1]);
      break;
    case K_DEPTHWISE:
      // ptrs: [x, w, bias, y]
      // ints: [N,H,W,C,R,S,sh,sw,pt,pl,Ho,Wo,act]
      launch_depthwise_conv(s, cp(0), cp(1), cp(2), p(3), int(I[0]),
                            int(I[1]), int(I[2]), int(I[3]), int(I[4]),
                            int(I[5]), int(I[6]), int(I[7]), int(I[8]),
                            int(I[9]), int(I[10]), int(I[11]),
                            int(I[12]));
      break;
    default:
      throw std::runtime_error("unknown call kind " +
                               std::to_string(c.kind));
  }
}

class ExecPlan {
 public:
  explicit ExecPlan(const std::vector<Call>& calls) : calls_(calls) {}

  void run() {
    hipStream_t s = c10::hip::getCurrentHIPStream().stream();
    for (const auto& c : calls_) launch_call(c, s);
    hipError_t e = hipGetLastError();
    if (e != hipSuccess)
      throw std::runtime_error(std::string("kernel launch failed: ") +
                               hipGetErrorString(e));
  }

  // capture the whole sequence into a hipGraph bound to fixed buffers;
  // subsequent run_graph() replays it (no per-kernel launch cost)
  void capture() {
    hipStream_t s = c10::hip::getCurrentHIPStream().stream();
    if (graph_exec_) return;
    hipGraph_t graph = nullptr;
    if (hipStreamBeginCapture(s, hipStreamCaptureModeThreadLocal) !=
        hipSuccess)
      throw std::runtime_error("hipStreamBeginCapture failed");
    for (const auto& c : calls_) launch_call(c, s);
    if (hipStreamEndCapture(s, &graph) != hipSuccess)
      throw std::runtime_error("hipStreamEndCapture failed");
    hipGraphExec_t ge = nullptr;
    if (hipGraphInstantiate(&ge, graph, nullptr, nullptr, 0) != hipSuccess) {
      hipGraphDestroy(graph);
      throw std::runtime_error("hipGraphInstantiate failed");
    }
    hipGraphDestroy(graph);
    graph_exec_ = ge;
  }

  bool has_graph() const { return graph_exec_ != nullptr; }

  void run_graph() {
    if (!graph_exec_) {
      run();
      return;
    }
    hipStream_t s = c10::hip::getCurrentHIPStream().stream();
    if (hipGraphLaunch(graph_exec_, s) != hipSuccess)
      throw std::runtime_error("hipGraphLaunch failed");
  }

  ~ExecPlan() {
    if (graph_exec_) hipGraphExecDestroy(graph_exec_);
  }

  // fastpath entry: replay (or launch eagerly) on an explicit stream
  void run_on(hipStream_t s) {
    if (graph_exec_) {
      if (hipGraphLaunch(graph_exec_, s) != hipSuccess)
        throw std::runtime_error("hipGraphLaunch failed");
      return;
    }
    for (const auto& c : calls_) launch_call(c, s);
    hipError_t e = hipGetLastError();
    if (e != hipSuccess)
      throw std::runtime_error(std::string("kernel launch failed: ") +
                               hipGetErrorString(e));
  }

  size_t n_calls() const { return calls_.size(); }

 private:
  std::vector<Call> calls_;
  hipGraphExec_t graph_exec_ = nullptr;
};

void fast_run_plan(void* plan, hipStream_t s) {
  reinterpret_cast<ExecPlan*>(plan)->run_on(s);
}

// Relocatable call template (plan-level context cache): the kernel
// calls of a (plan, bucket) with every pointer reduced to
// (tag, region index, offset). Instantiation rebases against a
// workspace base + region base table — one C++ call per context
// instead of rebuilding ~10^2 Python tuples per cold load.
struct PtrDesc {
  uint8_t tag;          // 0 = workspace+off, 1 = region[reg]+off, 2 = 0
  int32_t reg = 0;
  int64_t off = 0;
};

class CallTemplate {
 public:
  std::vector<Call> calls;                  // ptrs left empty
  std::vector<std::vector<PtrDesc>> descs;

  ExecPlan* instantiate(uintptr_t ws,
                        const std::vector<uintptr_t>& bases) const {
    std::vector<Call> out = calls;
    for (size_t i = 0; i < out.size(); ++i) {
      auto& ds = descs[i];
      auto& ptrs = out[i].ptrs;
      ptrs.resize(ds.size());
      for (size_t j = 0; j < ds.size(); ++j) {
        const PtrDesc& d = ds[j];
        if (d.tag == 0)
          ptrs[j] = intptr_t(ws) + d.off;
        else if (d.tag == 1)
          ptrs[j] = intptr_t(bases.at(size_t(d.reg))) + d.off;
        else
          ptrs[j] = 0;
      }
    }
    return new ExecPlan(out);
  }
};

}  // namespace tfsc

namespace py = pybind11;

void register_fastpath(py::module_& mod);
void register_frontend(py::module_& mod);
void register_rest_frontend(py::module_& mod);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  using tfsc::Call;
  using tfsc::ExecPlan;

  py::class_<ExecPlan>(mod, "ExecPlan")
      .def(py::init([](py::list calls) {
        std::vector<Call> cs;
        cs.reserve(calls.size());
        for (auto item : calls) {
          py::tuple t = item.cast<py::tuple>();
          Call c;
          c.kind = t[0].cast<int>();
          c.ptrs = t[1].cast<std::vector<intptr_t>>();
          c.ints = t[2].cast<std::vector<int64_t>>();
          c.floats = t[3].cast<std::vector<float>>();
          cs.push_back(std::move(c));
        }
        return new ExecPlan(cs);
      }))
      .def("run", &ExecPlan::run, py::call_guard<py::gil_scoped_release>())
      .def("capture", &ExecPlan::capture,
           py::call_guard<py::gil_scoped_release>())
      .def("run_graph", &ExecPlan::run_graph,
           py::call_guard<py::gil_scoped_release>())
      .def("has_graph", &ExecPlan::has_graph)
      .def("n_calls", &ExecPlan::n_calls)
      .def("ptr", [](ExecPlan& p) {
        return reinterpret_cast<uintptr_t>(&p);
      });

  mod.attr("K_ELT_UNARY") = int(tfsc::K_ELT_UNARY);
  mod.attr("K_ELT_BINARY") = int(tfsc::K_ELT_BINARY);
  mod.attr("K_BN_ACT") = int(tfsc::K_BN_ACT);
  mod.attr("K_SOFTMAX") = int(tfsc::K_SOFTMAX);
  mod.attr("K_LAYERNORM") = int(tfsc::K_LAYERNORM);
  mod.attr("K_MEAN_LAST") = int(tfsc::K_MEAN_LAST);
  mod.attr("K_MEAN_MID") = int(tfsc::K_MEAN_MID);
  mod.attr("K_POOL") = int(tfsc::K_POOL);
  mod.attr("K_TRANSPOSE") = int(tfsc::K_TRANSPOSE);
  mod.attr("K_GATHER") = int(tfsc::K_GATHER);
  mod.attr("K_PAD_NHWC") = int(tfsc::K_PAD_NHWC);
  mod.attr("K_GEMM") = int(tfsc::K_GEMM);
  mod.attr("K_BGEMM") = int(tfsc::K_BGEMM);
  mod.attr("K_IM2COL") = int(tfsc::K_IM2COL);
  mod.attr("K_CONV") = int(tfsc::K_CONV);
  mod.attr("K_PAD_LAST") = int(tfsc::K_PAD_LAST);
  mod.attr("K_ATTENTION") = int(tfsc::K_ATTENTION);
  mod.attr("K_DEPTHWISE") = int(tfsc::K_DEPTHWISE);
  mod.attr("K_GEMM_FP8") = int(tfsc::K_GEMM_FP8);
  mod.attr("K_QUANT_FP8") = int(tfsc::K_QUANT_FP8);
  mod.attr("K_CAST") = int(tfsc::K_CAST);
  mod.attr("K_ARGMAX_LAST") = int(tfsc::K_ARGMAX_LAST);
  mod.attr("K_SCATTER") = int(tfsc::K_SCATTER);

  py::class_<tfsc::CallTemplate>(mod, "CallTemplate")
      .def(py::init([](py::list calls) {
        auto* t = new tfsc::CallTemplate();
        t->calls.reserve(calls.size());
        t->descs.reserve(calls.size());
        for (auto item : calls) {
          py::tuple tu = item.cast<py::tuple>();
          Call c;
          c.kind = tu[0].cast<int>();
          std::vector<tfsc::PtrDesc> ds;
          for (auto dd : tu[1].cast<py::list>()) {
            py::tuple dt = dd.cast<py::tuple>();
            tfsc::PtrDesc d;
            d.tag = uint8_t(dt[0].cast<int>());
            if (d.tag == 0) {
              d.off = dt[1].cast<int64_t>();
            } else if (d.tag == 1) {
              d.reg = dt[1].cast<int32_t>();
              d.off = dt[2].cast<int64_t>();
            }
            ds.push_back(d);
          }
          c.ints = tu[2].cast<std::vector<int64_t>>();
          c.floats = tu[3].cast<std::vector<float>>();
          t->calls.push_back(std::move(c));
          t->descs.push_back(std::move(ds));
        }
        return t;
      }));

  mod.def("instantiate_plan",
          [](const tfsc::CallTemplate& t, uintptr_t ws,
             const std::vector<uintptr_t>& bases) {
            return std::unique_ptr<ExecPlan>(t.instantiate(ws, bases));
          });

  register_fastpath(mod);
  register_frontend(mod);
  register_rest_frontend(mod);

  // elementwise fn codes
  mod.attr("ELT_ADD") = int(tfsc::ELT_ADD);
  mod.attr("ELT_SUB") = int(tfsc::ELT_SUB);
  mod.attr("ELT_MUL") = int(tfsc::ELT_MUL);
  mod.attr("ELT_DIV") = int(tfsc::ELT_DIV);
  mod.attr("ELT_MAX") = int(tfsc::ELT_MAX);
  mod.attr("ELT_MIN") = int(tfsc::ELT_MIN);
  mod.attr("ELT_SQDIFF") = int(tfsc::ELT_SQDIFF);
  mod.attr("ELT_RELU") = int(tfsc::ELT_RELU);
  mod.attr("ELT_TANH") = int(tfsc::ELT_TANH);
  mod.attr("ELT_SIGMOID") = int(tfsc::ELT_SIGMOID);
  mod.attr("ELT_ERF") = int(tfsc::ELT_ERF);
  mod.attr("ELT_SQRT") = int(tfsc::ELT_SQRT);
  mod.attr("ELT_RSQRT") = int(tfsc::ELT_RSQRT);
  mod.attr("ELT_EXP") = int(tfsc::ELT_EXP);
  mod.attr("ELT_NEG") = int(tfsc::ELT_NEG);
  mod.attr("ELT_SQUARE") = int(tfsc::ELT_SQUARE);
  mod.attr("ELT_GELU") = int(tfsc::ELT_GELU);
  mod.attr("ELT_RELU6") = int(tfsc::ELT_RELU6);

  mod.attr("ACT_NONE") = int(tfsc::ACT_NONE);
  mod.attr("ACT_RELU") = int(tfsc::ACT_RELU);
  mod.attr("ACT_TANH") = int(tfsc::ACT_TANH);
  mod.attr("ACT_SIGMOID") = int(tfsc::ACT_SIGMOID);
  mod.attr("ACT_GELU") = int(tfsc::ACT_GELU);
  mod.attr("ACT_RELU6") = int(tfsc::ACT_RELU6);
}
