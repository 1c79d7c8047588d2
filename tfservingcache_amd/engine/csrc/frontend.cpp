// Native gRPC front-end for the cache tier.
//
// The Python grpc stack costs 35-80% of throughput at serving payloads
// (profiles/RESULTS.md): every request crosses the C-core -> Python
// boundary, holds the GIL for routing, and re-enters C++ for the fast
// path. This server replaces that hop: plain sockets + nghttp2 (HTTP/2
// framing, HPACK, flow control) + gRPC semantics implemented here
// (5-byte message framing, grpc-status trailers). Predict requests
// whose (model, version) is registered run ENTIRELY in C++ — peek the
// model_spec, look up the FastModel, execute, serialize — without ever
// taking the GIL. Everything else (cold loads, Classify/Regress/
// status/reload/session/health, fallback-worthy Predicts) calls a
// Python dispatcher under the GIL, preserving full wire compatibility.
//
// Reference scope note: the reference (Go sidecar) fronts TF Serving
// with grpc-go; this plays that role for the in-process engine.
#include <torch/extension.h>

#include <nghttp2/nghttp2.h>

#include <arpa/inet.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <sys/eventfd.h>
#include <sys/socket.h>
#include <unistd.h>

#include <atomic>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <map>
#include <memory>
#include <mutex>
#include <set>
#include <shared_mutex>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

#include "fastpath_api.h"

namespace py = pybind11;

namespace tfsc_fe {

static const char* kPredictPath =
    "/tensorflow.serving.PredictionService/Predict";

// gRPC status codes used here
enum { GRPC_OK = 0, GRPC_UNKNOWN = 2, GRPC_INVALID = 3,
       GRPC_UNIMPLEMENTED = 12, GRPC_INTERNAL = 13,
       GRPC_UNAVAILABLE = 14 };

struct Stream {
  std::string path;
  std::string body;
};

struct Response {
  int32_t stream_id = 0;
  int grpc_status = GRPC_OK;
  std::string message;            // grpc-message (errors)
  std::string body;               // framed gRPC payload
  size_t sent = 0;
  std::string status_str;         // storage for trailer values
  bool trailers_submitted = false;
};

class Frontend;

struct Conn {
  int fd = -1;
  int wake_fd = -1;
  nghttp2_session* sess = nullptr;
  Frontend* fe = nullptr;
  std::unordered_map<int32_t, Stream> streams;
  std::unordered_map<int32_t, std::unique_ptr<Response>> active;
  std::mutex out_mu;
  std::deque<std::unique_ptr<Response>> outbox;
  std::atomic<bool> dead{false};

  void enqueue(std::unique_ptr<Response> r) {
    // the wake write stays under out_mu: the conn thread sets `dead`
    // under the same mutex BEFORE closing wake_fd, so no writer can
    // race the close into a reused fd
    std::lock_guard<std::mutex> g(out_mu);
    if (dead.load()) return;
    outbox.push_back(std::move(r));
    uint64_t one = 1;
    ssize_t n = write(wake_fd, &one, sizeof(one));
    (void)n;
  }
};

struct Job {
  std::shared_ptr<Conn> conn;
  int32_t stream_id;
  std::string path;
  std::string body;
};

// ---------------------------------------------------------------------------
// nghttp2 callbacks
// ---------------------------------------------------------------------------
static int on_begin_headers(nghttp2_session*, const nghttp2_frame* frame,
                            void* user) {
  auto* conn = static_cast<Conn*>(user);
  if (frame->hd.type == NGHTTP2_HEADERS &&
      frame->headers.cat == NGHTTP2_HCAT_REQUEST)
    conn->streams[frame->hd.stream_id];
  return 0;
}

static int on_header(nghttp2_session*, const nghttp2_frame* frame,
                     const uint8_t* name, size_t namelen,
                     const uint8_t* value, size_t valuelen, uint8_t,
                     void* user) {
  auto* conn = static_cast<Conn*>(user);
  auto it = conn->streams.find(frame->hd.stream_id);
  if (it == conn->streams.end()) return 0;
  if (namelen == 5 && memcmp(name, ":path", 5) == 0)
    it->second.path.assign(reinterpret_cast<const char*>(value),
                           valuelen);
  return 0;
}

static int on_data_chunk(nghttp2_session*, uint8_t, int32_t stream_id,
                         const uint8_t* data, size_t len, void* user) {
  auto* conn = static_cast<Conn*>(user);
  auto it = conn->streams.find(stream_id);
  if (it == conn->streams.end()) return 0;
  if (it->second.body.size() + len > (256u << 20)) return 0;  // 256MB cap
  it->second.body.append(reinterpret_cast<const char*>(data), len);
  return 0;
}

static int on_stream_close(nghttp2_session*, int32_t stream_id, uint32_t,
                           void* user) {
  auto* conn = static_cast<Conn*>(user);
  conn->streams.erase(stream_id);
  conn->active.erase(stream_id);
  return 0;
}

static ssize_t response_read(nghttp2_session* sess, int32_t stream_id,
                             uint8_t* buf, size_t length,
                             uint32_t* data_flags, nghttp2_data_source*,
                             void* user) {
  auto* conn = static_cast<Conn*>(user);
  auto it = conn->active.find(stream_id);
  if (it == conn->active.end()) return NGHTTP2_ERR_CALLBACK_FAILURE;
  Response* r = it->second.get();
  size_t n = std::min(length, r->body.size() - r->sent);
  if (n) memcpy(buf, r->body.data() + r->sent, n);
  r->sent += n;
  if (r->sent == r->body.size()) {
    *data_flags |= NGHTTP2_DATA_FLAG_EOF | NGHTTP2_DATA_FLAG_NO_END_STREAM;
    if (!r->trailers_submitted) {
      r->trailers_submitted = true;
      r->status_str = std::to_string(r->grpc_status);
      std::vector<nghttp2_nv> trailers;
      auto nv = [](const char* k, const std::string& v) {
        return nghttp2_nv{
            const_cast<uint8_t*>(
                reinterpret_cast<const uint8_t*>(k)),
            const_cast<uint8_t*>(
                reinterpret_cast<const uint8_t*>(v.data())),
            strlen(k), v.size(), NGHTTP2_NV_FLAG_NONE};
      };
      trailers.push_back(nv("grpc-status", r->status_str));
      if (!r->message.empty())
        trailers.push_back(nv("grpc-message", r->message));
      nghttp2_submit_trailer(sess, stream_id, trailers.data(),
                             trailers.size());
    }
  }
  return ssize_t(n);
}

// ---------------------------------------------------------------------------
// Frontend
// ---------------------------------------------------------------------------
class Frontend {
 public:
  explicit Frontend(py::function fallback)
      : fallback_(std::move(fallback)) {}

  int start(int port, int n_workers) {
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
    for (int i = 0; i < std::max(1, n_workers); ++i)
      workers_.emplace_back([this] { worker_loop(); });
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
      std::lock_guard<std::mutex> g(conns_mu_);
      for (auto& c : conns_) {
        c->dead.store(true);
        uint64_t one = 1;
        ssize_t n = write(c->wake_fd, &one, sizeof(one));
        (void)n;
      }
    }
    jobs_cv_.notify_all();
    if (accept_thread_.joinable()) accept_thread_.join();
    for (auto& t : workers_)
      if (t.joinable()) t.join();
    workers_.clear();
    {
      std::lock_guard<std::mutex> g(threads_mu_);
      for (auto& t : conn_threads_)
        if (t.second.joinable()) t.second.join();
      conn_threads_.clear();
    }
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

  // stats for tests: requests served without touching Python
  long long native_hits() const { return native_hits_.load(); }
  long long fallback_calls() const { return fallback_calls_.load(); }

 private:
  // -- accept / connection loops ---------------------------------------
  void accept_loop() {
    while (!stopping_.load()) {
      int fd = accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) {
        if (stopping_.load()) return;
        continue;
      }
      int one = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      auto conn = std::make_shared<Conn>();
      conn->fd = fd;
      conn->fe = this;
      conn->wake_fd = eventfd(0, EFD_NONBLOCK);
      {
        std::lock_guard<std::mutex> g(conns_mu_);
        conns_.insert(conn);
      }
      {
        // reap threads of connections that have since closed (a
        // long-lived server must not accumulate unjoined stacks)
        std::lock_guard<std::mutex> g(threads_mu_);
        for (auto it = conn_threads_.begin();
             it != conn_threads_.end();) {
          if (it->first->dead.load() && it->second.joinable()) {
            it->second.join();
            it = conn_threads_.erase(it);
          } else {
            ++it;
          }
        }
        conn_threads_.emplace_back(
            conn, std::thread([this, conn] { conn_loop(conn); }));
      }
    }
  }

  void setup_session(const std::shared_ptr<Conn>& conn) {
    nghttp2_session_callbacks* cbs;
    nghttp2_session_callbacks_new(&cbs);
    nghttp2_session_callbacks_set_on_begin_headers_callback(
        cbs, on_begin_headers);
    nghttp2_session_callbacks_set_on_header_callback(cbs, on_header);
    nghttp2_session_callbacks_set_on_data_chunk_recv_callback(
        cbs, on_data_chunk);
    nghttp2_session_callbacks_set_on_stream_close_callback(
        cbs, on_stream_close);
    nghttp2_session_callbacks_set_on_frame_recv_callback(
        cbs, [](nghttp2_session*, const nghttp2_frame* frame,
                void* user) -> int {
          auto* c = static_cast<Conn*>(user);
          bool end = frame->hd.flags & NGHTTP2_FLAG_END_STREAM;
          if ((frame->hd.type == NGHTTP2_DATA ||
               frame->hd.type == NGHTTP2_HEADERS) && end) {
            auto it = c->streams.find(frame->hd.stream_id);
            if (it != c->streams.end())
              c->fe->dispatch(c, frame->hd.stream_id,
                              std::move(it->second));
            // body consumed; keep the entry until stream close
          }
          return 0;
        });
    nghttp2_session_server_new(&conn->sess, cbs, conn.get());
    nghttp2_session_callbacks_del(cbs);

    nghttp2_settings_entry settings[] = {
        {NGHTTP2_SETTINGS_MAX_CONCURRENT_STREAMS, 1024},
        {NGHTTP2_SETTINGS_INITIAL_WINDOW_SIZE, 8 << 20},
        {NGHTTP2_SETTINGS_MAX_FRAME_SIZE, 1 << 20},
    };
    nghttp2_submit_settings(conn->sess, NGHTTP2_FLAG_NONE, settings, 3);
    nghttp2_session_set_local_window_size(conn->sess, NGHTTP2_FLAG_NONE,
                                          0, 64 << 20);
  }

  void conn_loop(std::shared_ptr<Conn> conn) {
    setup_session(conn);
    std::vector<uint8_t> rbuf(1 << 16);
    while (!stopping_.load() && !conn->dead.load()) {
      flush_outbox(conn);
      if (send_pending(conn) < 0) break;
      pollfd fds[2] = {{conn->fd, POLLIN, 0},
                       {conn->wake_fd, POLLIN, 0}};
      int pr = poll(fds, 2, 500);
      if (pr < 0) break;
      if (fds[1].revents & POLLIN) {
        uint64_t junk;
        while (read(conn->wake_fd, &junk, sizeof(junk)) > 0) {
        }
      }
      if (fds[0].revents & (POLLIN | POLLHUP | POLLERR)) {
        ssize_t n = read(conn->fd, rbuf.data(), rbuf.size());
        if (n <= 0) break;
        ssize_t rv = nghttp2_session_mem_recv(conn->sess, rbuf.data(),
                                              size_t(n));
        if (rv < 0) break;
      }
    }
    {
      std::lock_guard<std::mutex> g(conn->out_mu);
      conn->dead.store(true);
    }
    {
      std::lock_guard<std::mutex> g(conns_mu_);
      conns_.erase(conn);
    }
    nghttp2_session_del(conn->sess);
    conn->sess = nullptr;
    close(conn->fd);
    close(conn->wake_fd);
  }

  void flush_outbox(const std::shared_ptr<Conn>& conn) {
    std::deque<std::unique_ptr<Response>> ready;
    {
      std::lock_guard<std::mutex> g(conn->out_mu);
      ready.swap(conn->outbox);
    }
    for (auto& r : ready) {
      int32_t sid = r->stream_id;
      std::string status = "200";
      nghttp2_nv hdrs[] = {
          {const_cast<uint8_t*>(
               reinterpret_cast<const uint8_t*>(":status")),
           const_cast<uint8_t*>(
               reinterpret_cast<const uint8_t*>(status.data())),
           7, status.size(), NGHTTP2_NV_FLAG_NONE},
          {const_cast<uint8_t*>(
               reinterpret_cast<const uint8_t*>("content-type")),
           const_cast<uint8_t*>(
               reinterpret_cast<const uint8_t*>("application/grpc")),
           12, 16, NGHTTP2_NV_FLAG_NONE},
      };
      conn->active[sid] = std::move(r);
      nghttp2_data_provider prd;
      prd.source.ptr = nullptr;
      prd.read_callback = response_read;
      if (nghttp2_submit_response(conn->sess, sid, hdrs, 2, &prd) != 0)
        conn->active.erase(sid);
    }
  }

  int send_pending(const std::shared_ptr<Conn>& conn) {
    while (true) {
      const uint8_t* out = nullptr;
      ssize_t n = nghttp2_session_mem_send(conn->sess, &out);
      if (n < 0) return -1;
      if (n == 0) return 0;
      size_t off = 0;
      while (off < size_t(n)) {
        ssize_t w = write(conn->fd, out + off, size_t(n) - off);
        if (w < 0) {
          if (errno == EINTR) continue;
          return -1;
        }
        off += size_t(w);
      }
    }
  }

  // -- request handling --------------------------------------------------
 public:
  void dispatch(Conn* raw, int32_t stream_id, Stream s) {
    std::shared_ptr<Conn> conn;
    {
      std::lock_guard<std::mutex> g(conns_mu_);
      for (auto& c : conns_)
        if (c.get() == raw) {
          conn = c;
          break;
        }
    }
    if (!conn) return;
    {
      std::lock_guard<std::mutex> g(jobs_mu_);
      jobs_.push_back(Job{std::move(conn), stream_id,
                          std::move(s.path), std::move(s.body)});
    }
    jobs_cv_.notify_one();
  }

 private:
  void worker_loop() {
    while (true) {
      Job job;
      {
        std::unique_lock<std::mutex> g(jobs_mu_);
        jobs_cv_.wait(g, [this] {
          return stopping_.load() || !jobs_.empty();
        });
        if (stopping_.load() && jobs_.empty()) return;
        job = std::move(jobs_.front());
        jobs_.pop_front();
      }
      auto resp = std::make_unique<Response>();
      resp->stream_id = job.stream_id;
      handle(job, resp.get());
      if (!job.conn->dead.load()) job.conn->enqueue(std::move(resp));
    }
  }

  static void frame_body(const std::string& payload, std::string* out) {
    out->reserve(payload.size() + 5);
    out->push_back('\0');
    uint32_t len = htonl(uint32_t(payload.size()));
    out->append(reinterpret_cast<const char*>(&len), 4);
    out->append(payload);
  }

  void handle(const Job& job, Response* resp) {
    // unwrap the gRPC 5-byte message frame
    if (job.body.size() < 5) {
      resp->grpc_status = GRPC_INVALID;
      resp->message = "truncated grpc frame";
      return;
    }
    if (job.body[0] != 0) {
      resp->grpc_status = GRPC_UNIMPLEMENTED;
      resp->message = "compressed messages not supported";
      return;
    }
    uint32_t mlen;
    memcpy(&mlen, job.body.data() + 1, 4);
    mlen = ntohl(mlen);
    if (size_t(mlen) + 5 > job.body.size()) {
      resp->grpc_status = GRPC_INVALID;
      resp->message = "grpc frame length mismatch";
      return;
    }
    const uint8_t* payload =
        reinterpret_cast<const uint8_t*>(job.body.data()) + 5;

    if (job.path == kPredictPath) {
      std::string name, label;
      long long version = 0;
      bool has_version = tfsc::peek_spec_raw(payload, mlen, &name,
                                             &version, &label);
      if (label.empty()) {
        tfsc::FastModel* fm = nullptr;
        {
          std::shared_lock<std::shared_mutex> g(reg_mu_);
          auto it = registry_.find(name);
          if (it != registry_.end() && !it->second.empty()) {
            if (has_version) {
              auto vit = it->second.find(version);
              if (vit != it->second.end()) fm = vit->second;
            } else {
              fm = it->second.rbegin()->second;   // latest = highest
            }
          }
        }
        if (fm != nullptr) {
          try {
            std::string out = tfsc::fastmodel_predict(fm, payload, mlen);
            native_hits_.fetch_add(1, std::memory_order_relaxed);
            frame_body(out, &resp->body);
            return;
          } catch (const tfsc::FastFallback&) {
            // fall through to Python
          } catch (const std::exception& e) {
            resp->grpc_status = GRPC_INTERNAL;
            resp->message = sanitize(e.what());
            return;
          }
        }
      }
    }
    call_python(job.path, payload, mlen, resp);
  }

  void call_python(const std::string& path, const uint8_t* payload,
                   size_t len, Response* resp) {
    fallback_calls_.fetch_add(1, std::memory_order_relaxed);
    py::gil_scoped_acquire gil;
    try {
      py::tuple r = fallback_(
          py::str(path),
          py::bytes(reinterpret_cast<const char*>(payload), len));
      int status = r[0].cast<int>();
      if (status != 0) {
        resp->grpc_status = status;
        resp->message = sanitize(r[1].cast<std::string>());
        return;
      }
      std::string body = r[2].cast<std::string>();
      frame_body(body, &resp->body);
    } catch (const std::exception& e) {
      resp->grpc_status = GRPC_INTERNAL;
      resp->message = sanitize(e.what());
    }
  }

  // grpc-message must be percent-encoded printable ASCII; keep simple
  static std::string sanitize(const std::string& s) {
    std::string out;
    out.reserve(s.size());
    for (char ch : s.substr(0, 512))
      out.push_back((ch >= 0x20 && ch < 0x7F && ch != '%') ? ch : '_');
    return out;
  }

  py::function fallback_;
  std::atomic<bool> stopping_{false};
  int listen_fd_ = -1;
  int bound_port_ = 0;
  std::thread accept_thread_;
  std::vector<std::thread> workers_;
  std::mutex threads_mu_;
  // shared_ptr key: the Conn must outlive the reaper's dead-check
  std::vector<std::pair<std::shared_ptr<Conn>, std::thread>>
      conn_threads_;
  std::mutex conns_mu_;
  std::set<std::shared_ptr<Conn>> conns_;
  std::mutex jobs_mu_;
  std::condition_variable jobs_cv_;
  std::deque<Job> jobs_;
  std::shared_mutex reg_mu_;
  std::map<std::string, std::map<long long, tfsc::FastModel*>> registry_;
  std::map<std::string, py::object> keepalive_;
  std::atomic<long long> native_hits_{0};
  std::atomic<long long> fallback_calls_{0};

 public:
  ~Frontend() {
    // destructor runs with the GIL held (pybind); workers may be
    // blocked acquiring it for a fallback call — release while joining
    if (PyGILState_Check()) {
      py::gil_scoped_release rel;
      stop();
    } else {
      stop();
    }
  }
};

}  // namespace tfsc_fe

void register_frontend(py::module_& mod) {
  using tfsc_fe::Frontend;
  py::class_<Frontend>(mod, "GrpcFrontend")
      .def(py::init<py::function>())
      .def("start", &Frontend::start, py::arg("port"),
           py::arg("workers") = 16,
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &Frontend::stop,
           py::call_guard<py::gil_scoped_release>())
      .def("port", &Frontend::port)
      .def("native_hits", &Frontend::native_hits)
      .def("fallback_calls", &Frontend::fallback_calls)
      .def("register_model",
           [](Frontend& fe, const std::string& name, long long version,
              uintptr_t fm_ptr, py::object keep) {
             fe.register_model(
                 name, version,
                 reinterpret_cast<tfsc::FastModel*>(fm_ptr),
                 std::move(keep));
           })
      .def("unregister_model", &Frontend::unregister_model);
}
