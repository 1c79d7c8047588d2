// Native REST front-end for the cache tier (HTTP/1.1 + JSON).
//
// Round-1 left the REST data path on Python/aiohttp while gRPC got the
// nghttp2 front-end; at b=1 REST paid the full Python tax. This server
// gives REST the same treatment: plain sockets + a hand-rolled
// HTTP/1.1 parser + a dense-JSON <-> protobuf bridge. A POST
// /v1/models/<m>[/versions/<v>]:predict whose model is registered and
// whose body is pure numeric instances/inputs runs ENTIRELY in C++:
// JSON numbers are parsed straight into the request tensor buffer, the
// existing FastModel path executes (incl. dynamic batching), and the
// response floats are serialized back to JSON — no GIL. Everything
// else (status/metadata/classify/regress/healthz/metrics, b64/string
// bodies, cold models) falls back to a Python dispatcher that reuses
// the same handlers as the aiohttp app, so the two REST servers are
// wire-identical.
//
// Reference parity: the URL grammar matches the reference's regex
// (pkg/tfservingproxy/tfservingproxy.go:24, case-insensitive, version
// optional like TF Serving itself).
#include <torch/extension.h>

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <atomic>
#include <cctype>
#include <chrono>
#include <cmath>
#include <cstring>
#include <map>
#include <memory>
#include <mutex>
#include <set>
#include <shared_mutex>
#include <string>
#include <thread>
#include <vector>

#include "fastpath_api.h"
#include "rest_parse.h"

namespace py = pybind11;

namespace tfsc_rest {

struct ParsedInput {
  std::vector<int64_t> dims;
  std::vector<float> f;
  std::vector<int32_t> i;
};

static std::string build_tensorproto(const ParsedInput& in, bool is_int) {
  std::string shape;
  for (int64_t d : in.dims) {
    std::string dim;
    w_tag(dim, 1, 0);
    w_varint(dim, uint64_t(d));
    w_len(shape, 2, dim);
  }
  std::string tp;
  w_tag(tp, 1, 0);
  w_varint(tp, is_int ? 3 : 1);          // DT_INT32 / DT_FLOAT
  w_len(tp, 2, shape);
  w_tag(tp, 4, 2);
  if (is_int) {
    w_varint(tp, in.i.size() * 4);
    tp.append(reinterpret_cast<const char*>(in.i.data()),
              in.i.size() * 4);
  } else {
    w_varint(tp, in.f.size() * 4);
    tp.append(reinterpret_cast<const char*>(in.f.data()),
              in.f.size() * 4);
  }
  return tp;
}

// ---------------------------------------------------------------------------
// JSON writer for the response
// ---------------------------------------------------------------------------
static void write_float(std::string& out, float v) {
  if (std::isnan(v)) { out += "NaN"; return; }         // TF emits bare
  if (std::isinf(v)) { out += v > 0 ? "Infinity" : "-Infinity"; return; }
  char buf[32];
  int n = snprintf(buf, sizeof(buf), "%.8g", double(v));
  out.append(buf, size_t(n));
}

// write values[off .. off+prod(dims[d:])) as nested arrays
static void write_nested(std::string& out, const float* vals,
                         const std::vector<int64_t>& dims, size_t d,
                         size_t& off) {
  if (d == dims.size()) {
    write_float(out, vals[off++]);
    return;
  }
  out.push_back('[');
  for (int64_t i = 0; i < dims[d]; ++i) {
    if (i) out.push_back(',');
    write_nested(out, vals, dims, d + 1, off);
  }
  out.push_back(']');
}

// ---------------------------------------------------------------------------
// HTTP/1.1 connection handling
// ---------------------------------------------------------------------------
struct HttpRequest {
  std::string method;
  std::string path;
  std::string body;
  bool keep_alive = true;
};

class RestFrontend {
 public:
  explicit RestFrontend(py::function fallback)
      : fallback_(std::move(fallback)) {}

  int start(int port) {
    listen_fd_ = socket(AF_INET, SOCK_STREAM, 0);
    if (listen_fd_ < 0) throw std::runtime_error("socket() failed");
    int one = 1;
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_addr.s_addr = htonl(INADDR_ANY);
    addr.sin_port = htons(uint16_t(port));
    if (bind(listen_fd_, reinterpret_cast<sockaddr*>(&addr),
             sizeof(addr)) != 0) {
      close(listen_fd_);
      listen_fd_ = -1;
      throw std::runtime_error("bind() failed on port " +
                               std::to_string(port));
    }
    socklen_t alen = sizeof(addr);
    getsockname(listen_fd_, reinterpret_cast<sockaddr*>(&addr), &alen);
    bound_port_ = ntohs(addr.sin_port);
    listen(listen_fd_, 128);
    stopping_.store(false);
    accept_thread_ = std::thread([this] { accept_loop(); });
    return bound_port_;
  }

  void stop() {
    stopping_.store(true);
    if (listen_fd_ >= 0) {
      shutdown(listen_fd_, SHUT_RDWR);
      close(listen_fd_);
      listen_fd_ = -1;
    }
    {
      std::lock_guard<std::mutex> g(fds_mu_);
      for (int fd : conn_fds_) shutdown(fd, SHUT_RDWR);
    }
    if (accept_thread_.joinable()) accept_thread_.join();
    std::lock_guard<std::mutex> g(threads_mu_);
    for (auto& t : conn_threads_)
      if (t.second.joinable()) t.second.join();
    conn_threads_.clear();
  }

  void register_model(const std::string& name, long long version,
                      tfsc::FastModel* fm, py::object keep) {
    std::unique_lock<std::shared_mutex> g(reg_mu_);
    registry_[name][version] = fm;
    keepalive_[name + "##" + std::to_string(version)] = std::move(keep);
  }

  void unregister_model(const std::string& name, long long version) {
    std::unique_lock<std::shared_mutex> g(reg_mu_);
    auto it = registry_.find(name);
    if (it != registry_.end()) {
      it->second.erase(version);
      if (it->second.empty()) registry_.erase(it);
    }
    keepalive_.erase(name + "##" + std::to_string(version));
  }

  int port() const { return bound_port_; }
  long long native_hits() const { return native_hits_.load(); }
  long long fallback_calls() const { return fallback_calls_.load(); }

 private:
  void accept_loop() {
    while (!stopping_.load()) {
      int fd = accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) {
        if (stopping_.load()) return;
        continue;
      }
      int one = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      {
        std::lock_guard<std::mutex> g(fds_mu_);
        conn_fds_.insert(fd);
      }
      std::lock_guard<std::mutex> g(threads_mu_);
      // reap finished connection threads so a long-lived server does
      // not accumulate unjoined stacks under connection churn
      for (auto it = conn_threads_.begin(); it != conn_threads_.end();) {
        if (it->first->load() && it->second.joinable()) {
          it->second.join();
          it = conn_threads_.erase(it);
        } else {
          ++it;
        }
      }
      auto done = std::make_shared<std::atomic<bool>>(false);
      conn_threads_.emplace_back(done, std::thread([this, fd, done] {
        conn_loop(fd);
        {
          std::lock_guard<std::mutex> g2(fds_mu_);
          conn_fds_.erase(fd);
        }
        done->store(true);
      }));
    }
  }

  // returns false on EOF/parse failure (close connection)
  bool read_request(int fd, std::string& buf, HttpRequest* req) {
    size_t hdr_end;
    while ((hdr_end = buf.find("\r\n\r\n")) == std::string::npos) {
      if (buf.size() > (1u << 20)) return false;     // header cap 1MB
      char tmp[1 << 14];
      ssize_t n = read(fd, tmp, sizeof(tmp));
      if (n <= 0) return false;
      buf.append(tmp, size_t(n));
    }
    // request line
    size_t line_end = buf.find("\r\n");
    {
      const std::string line = buf.substr(0, line_end);
      size_t sp1 = line.find(' ');
      size_t sp2 = line.find(' ', sp1 + 1);
      if (sp1 == std::string::npos || sp2 == std::string::npos)
        return false;
      req->method = line.substr(0, sp1);
      req->path = line.substr(sp1 + 1, sp2 - sp1 - 1);
    }
    size_t content_len = 0;
    bool chunked = false;
    req->keep_alive = true;
    for (size_t pos = line_end + 2; pos < hdr_end;) {
      size_t eol = buf.find("\r\n", pos);
      if (eol == std::string::npos || eol > hdr_end) break;
      std::string line = buf.substr(pos, eol - pos);
      pos = eol + 2;
      size_t colon = line.find(':');
      if (colon == std::string::npos) continue;
      std::string key = line.substr(0, colon);
      for (auto& ch : key) ch = char(tolower(ch));
      std::string val = line.substr(colon + 1);
      while (!val.empty() && val.front() == ' ') val.erase(0, 1);
      if (key == "content-length") content_len = size_t(atoll(val.c_str()));
      else if (key == "connection") {
        for (auto& ch : val) ch = char(tolower(ch));
        if (val == "close") req->keep_alive = false;
      } else if (key == "transfer-encoding") {
        chunked = true;
      }
    }
    if (chunked || content_len > (256u << 20)) return false;
    size_t body_start = hdr_end + 4;
    while (buf.size() < body_start + content_len) {
      char tmp[1 << 16];
      ssize_t n = read(fd, tmp, sizeof(tmp));
      if (n <= 0) return false;
      buf.append(tmp, size_t(n));
    }
    req->body = buf.substr(body_start, content_len);
    buf.erase(0, body_start + content_len);
    return true;
  }

  static bool write_all(int fd, const std::string& s) {
    size_t off = 0;
    while (off < s.size()) {
      ssize_t n = write(fd, s.data() + off, s.size() - off);
      if (n < 0) {
        if (errno == EINTR) continue;
        return false;
      }
      off += size_t(n);
    }
    return true;
  }

  void conn_loop(int fd) {
    std::string buf;
    while (!stopping_.load()) {
      HttpRequest req;
      if (!read_request(fd, buf, &req)) break;
      int status = 200;
      std::string ctype = "application/json";
      std::string body;
      handle(req, &status, &ctype, &body);
      std::string resp;
      resp.reserve(body.size() + 128);
      resp += "HTTP/1.1 " + std::to_string(status) +
              (status == 200 ? " OK" : " Error") + "\r\n";
      resp += "Content-Type: " + ctype + "\r\n";
      resp += "Content-Length: " + std::to_string(body.size()) + "\r\n";
      if (!req.keep_alive) resp += "Connection: close\r\n";
      resp += "\r\n";
      resp += body;
      if (!write_all(fd, resp)) break;
      if (!req.keep_alive) break;
    }
    close(fd);
  }

  // -- URL parse (reference regex semantics) -----------------------------
  // /v1/models/<name>[/versions/<digits>]:predict  (case-insensitive
  // path components; name may not contain '/' or ':')
  static bool parse_predict_url(const std::string& path,
                                std::string* name, long long* version,
                                bool* has_version) {
    auto ieq = [](const std::string& s, size_t off, const char* lit) {
      size_t n = strlen(lit);
      if (off + n > s.size()) return false;
      for (size_t i = 0; i < n; ++i)
        if (tolower(s[off + i]) != lit[i]) return false;
      return true;
    };
    if (!ieq(path, 0, "/v1/models/")) return false;
    size_t pos = 11;
    size_t end = path.find_first_of("/:", pos);
    if (end == std::string::npos || end == pos) return false;
    *name = path.substr(pos, end - pos);
    *has_version = false;
    *version = 0;
    pos = end;
    if (path[pos] == '/') {
      if (!ieq(path, pos, "/versions/")) return false;
      pos += 10;
      size_t vend = pos;
      while (vend < path.size() && isdigit(path[vend])) ++vend;
      if (vend == pos) return false;
      *version = atoll(path.substr(pos, vend - pos).c_str());
      *has_version = true;
      pos = vend;
    }
    return ieq(path, pos, ":predict") && pos + 8 == path.size();
  }

  void handle(const HttpRequest& req, int* status, std::string* ctype,
              std::string* body) {
    std::string name;
    long long version = 0;
    bool has_version = false;
    if (req.method == "POST" &&
        parse_predict_url(req.path, &name, &version, &has_version)) {
      tfsc::FastModel* fm = nullptr;
      {
        std::shared_lock<std::shared_mutex> g(reg_mu_);
        auto it = registry_.find(name);
        if (it != registry_.end() && !it->second.empty()) {
          if (has_version) {
            auto vit = it->second.find(version);
            if (vit != it->second.end()) fm = vit->second;
          } else {
            fm = it->second.rbegin()->second;
          }
        }
      }
      if (fm != nullptr) {
        try {
          if (fast_predict(fm, name, version, has_version, req.body,
                           body)) {
            native_hits_.fetch_add(1, std::memory_order_relaxed);
            *status = 200;
            return;
          }
        } catch (const RestFallback&) {
          // python path below
        } catch (const tfsc::FastFallback&) {
        } catch (const std::exception& e) {
          *status = 500;
          *body = std::string("{\"error\": \"") + e.what() + "\"}";
          return;
        }
      }
    }
    call_python(req, status, ctype, body);
  }

  // JSON -> PredictRequest proto -> FastModel -> PredictResponse proto
  // -> JSON. Returns false (or throws RestFallback/FastFallback) when
  // the request needs Python.
  bool fast_predict(tfsc::FastModel* fm, const std::string& name,
                    long long version, bool has_version,
                    const std::string& body, std::string* out) {
    auto specs = tfsc::fastmodel_input_specs(fm);
    if (specs.empty()) return false;
    JParser j{body.data(), body.data() + body.size()};
    j.expect('{');
    bool row_format = false;
    std::map<std::string, ParsedInput> parsed;
    bool saw_payload = false;
    while (true) {
      std::string key = j.string();
      j.expect(':');
      if (key == "signature_name") {
        std::string sig = j.string();
        if (!sig.empty() && sig != "serving_default")
          throw RestFallback("signature");
      } else if (key == "instances" || key == "inputs") {
        if (saw_payload) throw RestFallback("both payload keys");
        saw_payload = true;
        row_format = (key == "instances");
        j.ws();
        if (row_format && j.p < j.end && *j.p == '[') {
          // peek: list of objects (named rows) or pure numeric array
          const char* save = j.p;
          ++j.p;
          j.ws();
          if (j.p < j.end && *j.p == '{') {
            parse_named_rows(j, specs, parsed);
          } else {
            j.p = save;
            parse_anon(j, specs, parsed);
          }
        } else {
          j.ws();
          if (j.p < j.end && *j.p == '{') {
            // columnar named: {"alias": nested, ...}
            ++j.p;
            while (true) {
              std::string alias = j.string();
              j.expect(':');
              const tfsc::FastIOSpec* sp = find_spec(specs, alias);
              if (!sp) throw RestFallback("unknown input");
              ParsedInput& pi = parsed[alias];
              parse_dense(j, 0, pi.dims, pi.f, pi.i, sp->is_int);
              j.ws();
              if (j.p < j.end && *j.p == ',') { ++j.p; continue; }
              break;
            }
            j.expect('}');
          } else {
            parse_anon(j, specs, parsed);
          }
        }
      } else {
        throw RestFallback("unknown key");
      }
      j.ws();
      if (j.p < j.end && *j.p == ',') { ++j.p; continue; }
      break;
    }
    j.expect('}');
    if (!saw_payload || parsed.empty()) throw RestFallback("no payload");
    for (auto& kv : parsed) check_dims_complete(kv.second.dims);

    // build the PredictRequest
    std::string reqpb;
    {
      std::string spec;
      w_tag(spec, 1, 2);
      w_varint(spec, name.size());
      spec += name;
      if (has_version) {
        std::string ver;
        w_tag(ver, 1, 0);
        w_varint(ver, uint64_t(version));
        w_len(spec, 2, ver);
      }
      w_len(reqpb, 1, spec);
    }
    for (auto& kv : parsed) {
      const tfsc::FastIOSpec* sp = find_spec(specs, kv.first);
      std::string entry;
      w_tag(entry, 1, 2);
      w_varint(entry, kv.first.size());
      entry += kv.first;
      w_len(entry, 2, build_tensorproto(kv.second, sp && sp->is_int));
      w_len(reqpb, 2, entry);
    }

    std::string resp = tfsc::fastmodel_predict(
        fm, reinterpret_cast<const uint8_t*>(reqpb.data()), reqpb.size());
    auto outs = parse_response(resp);
    if (outs.empty()) return false;
    render_json(outs, row_format, out);
    return true;
  }

  static const tfsc::FastIOSpec* find_spec(
      const std::vector<tfsc::FastIOSpec>& specs,
      const std::string& alias) {
    for (auto& s : specs)
      if (s.alias == alias) return &s;
    return nullptr;
  }

  // {"instances": <numeric array>} with the single input anonymous;
  // the array's first dim is the batch
  static void parse_anon(JParser& j,
                         const std::vector<tfsc::FastIOSpec>& specs,
                         std::map<std::string, ParsedInput>& parsed) {
    if (specs.size() != 1) throw RestFallback("anonymous multi-input");
    ParsedInput& pi = parsed[specs[0].alias];
    parse_dense(j, 0, pi.dims, pi.f, pi.i, specs[0].is_int);
    if (pi.dims.empty()) throw RestFallback("scalar instances");
  }

  // {"instances": [{"a": ..., "b": ...}, ...]} — j.p is just past the
  // first '{'s opening '[' and pointing at '{'
  static void parse_named_rows(JParser& j,
                               const std::vector<tfsc::FastIOSpec>& specs,
                               std::map<std::string, ParsedInput>& parsed) {
    int64_t rows = 0;
    while (true) {
      j.expect('{');
      std::map<std::string, std::vector<int64_t>> row_dims;
      while (true) {
        std::string alias = j.string();
        j.expect(':');
        const tfsc::FastIOSpec* sp = find_spec(specs, alias);
        if (!sp) throw RestFallback("unknown input");
        ParsedInput& pi = parsed[alias];
        std::vector<int64_t> dims;     // dims of ONE row (tail)
        auto& rd = row_dims[alias];
        if (rows == 0) {
          parse_dense(j, 0, rd, pi.f, pi.i, sp->is_int);
        } else {
          // enforce same tail dims as row 0 by parsing with the known
          // dims of a single row
          std::vector<int64_t> expect_tail = pi.dims;  // tail-only here
          parse_dense(j, 0, dims, pi.f, pi.i, sp->is_int);
          if (dims != expect_tail) throw RestFallback("ragged rows");
        }
        (void)dims;
        j.ws();
        if (j.p < j.end && *j.p == ',') { ++j.p; continue; }
        break;
      }
      j.expect('}');
      if (rows == 0) {
        // pi.dims currently holds the TAIL dims for each alias
        for (auto& kv : row_dims) parsed[kv.first].dims = kv.second;
        if (row_dims.size() != specs.size())
          throw RestFallback("missing inputs in row");
      }
      ++rows;
      j.ws();
      if (j.p < j.end && *j.p == ',') { ++j.p; continue; }
      break;
    }
    j.expect(']');
    // prepend the batch dim
    for (auto& kv : parsed) {
      std::vector<int64_t> full;
      full.push_back(rows);
      for (int64_t d : kv.second.dims) full.push_back(d);
      kv.second.dims = std::move(full);
    }
  }

  static void render_json(const std::vector<OutTensor>& outs,
                          bool row_format, std::string* out) {
    size_t total_elems = 0;
    for (auto& t : outs) total_elems += t.nbytes / 4;
    out->reserve(total_elems * 12 + 64);
    if (!row_format) {
      // columnar: {"outputs": ...}
      out->append("{\"outputs\": ");
      if (outs.size() == 1) {
        size_t off = 0;
        write_nested(*out, reinterpret_cast<const float*>(outs[0].data),
                     outs[0].dims, 0, off);
      } else {
        out->push_back('{');
        for (size_t i = 0; i < outs.size(); ++i) {
          if (i) out->push_back(',');
          out->push_back('"');
          out->append(outs[i].alias);
          out->append("\": ");
          size_t off = 0;
          write_nested(*out, reinterpret_cast<const float*>(outs[i].data),
                       outs[i].dims, 0, off);
        }
        out->push_back('}');
      }
      out->push_back('}');
      return;
    }
    // row: {"predictions": [...]}
    out->append("{\"predictions\": ");
    if (outs.size() == 1) {
      size_t off = 0;
      write_nested(*out, reinterpret_cast<const float*>(outs[0].data),
                   outs[0].dims, 0, off);
    } else {
      int64_t rows = outs[0].dims.empty() ? 0 : outs[0].dims[0];
      std::vector<size_t> offs(outs.size(), 0);
      out->push_back('[');
      for (int64_t r = 0; r < rows; ++r) {
        if (r) out->push_back(',');
        out->push_back('{');
        for (size_t i = 0; i < outs.size(); ++i) {
          if (i) out->push_back(',');
          out->push_back('"');
          out->append(outs[i].alias);
          out->append("\": ");
          std::vector<int64_t> tail(outs[i].dims.begin() + 1,
                                    outs[i].dims.end());
          write_nested(*out,
                       reinterpret_cast<const float*>(outs[i].data),
                       tail, 0, offs[i]);
        }
        out->push_back('}');
      }
      out->push_back(']');
    }
    out->push_back('}');
  }

  void call_python(const HttpRequest& req, int* status,
                   std::string* ctype, std::string* body) {
    fallback_calls_.fetch_add(1, std::memory_order_relaxed);
    py::gil_scoped_acquire gil;
    try {
      py::tuple r = fallback_(py::str(req.method), py::str(req.path),
                              py::bytes(req.body));
      *status = r[0].cast<int>();
      *ctype = r[1].cast<std::string>();
      *body = r[2].cast<std::string>();
    } catch (const std::exception& e) {
      *status = 500;
      *ctype = "application/json";
      *body = std::string("{\"error\": \"dispatcher failed\"}");
      (void)e;
    }
  }

  py::function fallback_;
  std::atomic<bool> stopping_{false};
  int listen_fd_ = -1;
  int bound_port_ = 0;
  std::thread accept_thread_;
  std::mutex threads_mu_;
  std::vector<std::pair<std::shared_ptr<std::atomic<bool>>, std::thread>>
      conn_threads_;
  std::mutex fds_mu_;
  std::set<int> conn_fds_;
  std::shared_mutex reg_mu_;
  std::map<std::string, std::map<long long, tfsc::FastModel*>> registry_;
  std::map<std::string, py::object> keepalive_;
  std::atomic<long long> native_hits_{0};
  std::atomic<long long> fallback_calls_{0};

 public:
  ~RestFrontend() {
    if (PyGILState_Check()) {
      py::gil_scoped_release rel;
      stop();
    } else {
      stop();
    }
  }
};

}  // namespace tfsc_rest

void register_rest_frontend(py::module_& mod) {
  using tfsc_rest::RestFrontend;
  py::class_<RestFrontend>(mod, "RestFrontendNative")
      .def(py::init<py::function>())
      .def("start", &RestFrontend::start, py::arg("port"),
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &RestFrontend::stop,
           py::call_guard<py::gil_scoped_release>())
      .def("port", &RestFrontend::port)
      .def("native_hits", &RestFrontend::native_hits)
      .def("fallback_calls", &RestFrontend::fallback_calls)
      .def("register_model",
           [](RestFrontend& fe, const std::string& name,
              long long version, uintptr_t fm_ptr, py::object keep) {
             fe.register_model(
                 name, version,
                 reinterpret_cast<tfsc::FastModel*>(fm_ptr),
                 std::move(keep));
           })
      .def("unregister_model", &RestFrontend::unregister_model);

  // test-only: run the dense-JSON array parser and return (dims, data)
  mod.def("_rest_parse_probe", [](py::bytes body, bool is_int) {
    std::string b = body;
    tfsc_rest::JParser j{b.data(), b.data() + b.size()};
    std::vector<int64_t> dims;
    std::vector<float> f;
    std::vector<int32_t> iv;
    tfsc_rest::parse_dense(j, 0, dims, f, iv, is_int);
    tfsc_rest::check_dims_complete(dims);
    py::list pd;
    for (auto d : dims) pd.append(d);
    py::list pv;
    if (is_int)
      for (auto v : iv) pv.append(v);
    else
      for (auto v : f) pv.append(v);
    return py::make_tuple(pd, pv);
  });

  // test-only: time the raw parse (no python list conversion)
  mod.def("_rest_parse_bench", [](py::bytes body, int iters) {
    std::string b = body;
    double best = 1e30;
    for (int i = 0; i < iters; ++i) {
      std::vector<int64_t> dims;
      std::vector<float> f;
      std::vector<int32_t> iv;
      auto t0 = std::chrono::steady_clock::now();
      tfsc_rest::JParser j{b.data(), b.data() + b.size()};
      tfsc_rest::parse_dense(j, 0, dims, f, iv, false);
      auto dt = std::chrono::duration<double>(
          std::chrono::steady_clock::now() - t0).count();
      if (dt < best) best = dt;
    }
    return best;
  });
}
