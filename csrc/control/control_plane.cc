// Native TCP control plane — capability C1.
//
// The reference starts a gRPC server (Master/Worker services) in every
// process (tensorflow/core/distributed_runtime/rpc/grpc_server_lib.cc).
// rocps's equivalent: one threaded TCP server per process speaking the
// length-prefixed-JSON protocol of rocps/cluster.py, with the
// cluster-coordination ops (kv store, blocking gets, named barriers,
// atomic counters, membership/heartbeats, shutdown) served natively —
// no GIL on the coordination path while workers train. Unknown ops are
// forwarded to a Python fallback handler (async-PS shard service).
//
// Wire format (matches rocps/cluster.py):
//   <u64 total_len><u64 json_len><json bytes><payload bytes>
//
// JSON handling is a minimal self-contained span parser: top-level
// object fields are located as raw spans; scalar fields are decoded;
// "value" spans are stored/echoed verbatim.

#include <pybind11/functional.h>
#include <pybind11/pybind11.h>

#include <arpa/inet.h>
#include <execinfo.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstring>
#include <map>
#include <mutex>
#include <set>
#include <string>
#include <thread>
#include <vector>

namespace py = pybind11;

namespace {

// ---------------- minimal JSON span utilities ------------------------------
struct JsonField {
  std::string raw;  // value as raw JSON text
};

const char* skip_ws(const char* p, const char* e) {
  while (p < e && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) p++;
  return p;
}

// returns end of the JSON value starting at p (assumes valid JSON)
const char* skip_value(const char* p, const char* e) {
  p = skip_ws(p, e);
  if (p >= e) return p;
  if (*p == '"') {
    p++;
    while (p < e) {
      if (*p == '\\') p += 2;
      else if (*p == '"') return p + 1;
      else p++;
    }
    return e;
  }
  if (*p == '{' || *p == '[') {
    char open = *p, close = (*p == '{') ? '}' : ']';
    int depth = 0;
    bool in_str = false;
    while (p < e) {
      if (in_str) {
        if (*p == '\\') p++;
        else if (*p == '"') in_str = false;
      } else {
        if (*p == '"') in_str = true;
        else if (*p == open) depth++;
        else if (*p == close) {
          depth--;
          if (depth == 0) return p + 1;
        }
      }
      p++;
    }
    return e;
  }
  while (p < e && *p != ',' && *p != '}' && *p != ']' && *p != ' '
         && *p != '\n' && *p != '\t' && *p != '\r') p++;
  return p;
}

std::string json_unescape(const std::string& raw) {
  // raw includes surrounding quotes
  std::string out;
  for (size_t i = 1; i + 1 < raw.size(); i++) {
    char c = raw[i];
    if (c == '\\' && i + 2 < raw.size() + 1) {
      char n = raw[++i];
      switch (n) {
        case 'n': out += '\n'; break;
        case 't': out += '\t'; break;
        case 'r': out += '\r'; break;
        case '"': out += '"'; break;
        case '\\': out += '\\'; break;
        case '/': out += '/'; break;
        default: out += n;
      }
    } else {
      out += c;
    }
  }
  return out;
}

std::string json_escape(const std::string& s) {
  std::string out = "\"";
  for (char c : s) {
    switch (c) {
      case '"': out += "\\\""; break;
      case '\\': out += "\\\\"; break;
      case '\n': out += "\\n"; break;
      case '\t': out += "\\t"; break;
      case '\r': out += "\\r"; break;
      default: out += c;
    }
  }
  return out + "\"";
}

// parse top-level object into field -> raw-value map
bool parse_object(const std::string& s,
                  std::map<std::string, JsonField>* out) {
  const char* p = s.data();
  const char* e = p + s.size();
  p = skip_ws(p, e);
  if (p >= e || *p != '{') return false;
  p++;
  while (true) {
    p = skip_ws(p, e);
    if (p < e && *p == '}') return true;
    if (p >= e || *p != '"') return false;
    const char* kend = skip_value(p, e);
    std::string key = json_unescape(std::string(p, kend));
    p = skip_ws(kend, e);
    if (p >= e || *p != ':') return false;
    p = skip_ws(p + 1, e);
    const char* vend = skip_value(p, e);
    (*out)[key] = JsonField{std::string(p, vend)};
    p = skip_ws(vend, e);
    if (p < e && *p == ',') p++;
    else if (p < e && *p == '}') return true;
    else if (p >= e) return false;
  }
}

double to_number(const JsonField& f, double dflt = 0.0) {
  try {
    return std::stod(f.raw);
  } catch (...) {
    return dflt;
  }
}

std::string to_string_field(const JsonField& f) {
  if (!f.raw.empty() && f.raw[0] == '"') return json_unescape(f.raw);
  return f.raw;
}

// ---------------- socket helpers -------------------------------------------
bool send_all(int fd, const char* buf, size_t n) {
  while (n > 0) {
    ssize_t w = ::send(fd, buf, n, MSG_NOSIGNAL);
    if (w <= 0) return false;
    buf += w;
    n -= (size_t)w;
  }
  return true;
}

bool recv_all(int fd, char* buf, size_t n) {
  while (n > 0) {
    ssize_t r = ::recv(fd, buf, n, 0);
    if (r <= 0) return false;
    buf += r;
    n -= (size_t)r;
  }
  return true;
}

bool send_msg(int fd, const std::string& head, const std::string& payload) {
  uint64_t jlen = head.size();
  uint64_t total = 8 + jlen + payload.size();
  char hdr[16];
  memcpy(hdr, &total, 8);
  memcpy(hdr + 8, &jlen, 8);
  return send_all(fd, hdr, 16) && send_all(fd, head.data(), head.size()) &&
         (payload.empty() || send_all(fd, payload.data(), payload.size()));
}

bool recv_msg(int fd, std::string* head, std::string* payload) {
  char hdr[16];
  if (!recv_all(fd, hdr, 16)) return false;
  uint64_t total, jlen;
  memcpy(&total, hdr, 8);
  memcpy(&jlen, hdr + 8, 8);
  if (total > (1ull << 30) || jlen > total - 8) return false;
  head->resize(jlen);
  if (!recv_all(fd, head->data(), jlen)) return false;
  uint64_t plen = total - 8 - jlen;
  payload->resize(plen);
  if (plen && !recv_all(fd, payload->data(), plen)) return false;
  return true;
}

double now_s() {
  return std::chrono::duration<double>(
             std::chrono::system_clock::now().time_since_epoch())
      .count();
}

// ---------------- server ---------------------------------------------------
class ControlPlaneServer {
 public:
  ControlPlaneServer(int port, bool is_master, const std::string& job,
                     long task)
      : port_(port), is_master_(is_master), job_(job), task_(task) {}

  ~ControlPlaneServer() { stop(); }

  int start(const std::string& bind_host) {
    listen_fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
    if (listen_fd_ < 0) throw std::runtime_error("socket() failed");
    int one = 1;
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons((uint16_t)port_);
    addr.sin_addr.s_addr =
        bind_host.empty() ? INADDR_ANY : inet_addr(bind_host.c_str());
    if (bind(listen_fd_, (sockaddr*)&addr, sizeof(addr)) != 0)
      throw std::runtime_error("bind() failed on port " +
                               std::to_string(port_));
    socklen_t alen = sizeof(addr);
    getsockname(listen_fd_, (sockaddr*)&addr, &alen);
    port_ = ntohs(addr.sin_port);
    if (listen(listen_fd_, 128) != 0)
      throw std::runtime_error("listen() failed");
    running_ = true;
    accept_thread_ = std::thread([this] { accept_loop(); });
    return port_;
  }

  void set_fallback(py::function fn) {
    std::lock_guard<std::mutex> g(py_mu_);
    fallback_ = fn;
    has_fallback_ = true;
  }

  void stop() {
    if (!running_.exchange(false)) return;
    ::shutdown(listen_fd_, SHUT_RDWR);
    ::close(listen_fd_);
    {
      std::lock_guard<std::mutex> g(mu_);
      shutdown_flag_ = true;
    }
    cv_.notify_all();
    if (accept_thread_.joinable()) accept_thread_.join();
    // sweep until empty: a connection accepted in the instant before the
    // listen socket closed may have spawned a thread after an earlier
    // sweep — destroying a joinable std::thread calls std::terminate.
    while (true) {
      std::vector<std::thread> conns;
      {
        std::lock_guard<std::mutex> g(conn_mu_);
        if (conn_threads_.empty()) break;
        conns.swap(conn_threads_);
        for (int fd : conn_fds_) ::shutdown(fd, SHUT_RDWR);
      }
      for (auto& t : conns)
        if (t.joinable()) t.join();
    }
  }

  bool shutdown_requested() {
    std::lock_guard<std::mutex> g(mu_);
    return shutdown_flag_;
  }

  bool wait_shutdown(double timeout) {
    std::unique_lock<std::mutex> lk(mu_);
    if (timeout <= 0) {
      cv_.wait(lk, [this] { return shutdown_flag_; });
      return true;
    }
    return cv_.wait_for(lk, std::chrono::duration<double>(timeout),
                        [this] { return shutdown_flag_; });
  }

  int port() const { return port_; }

 private:
  void accept_loop() {
    while (running_) {
      int fd = ::accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) {
        if (!running_) break;
        continue;
      }
      int one = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      std::lock_guard<std::mutex> g(conn_mu_);
      if (!running_) {  // raced with stop(): do not spawn past the sweep
        ::close(fd);
        break;
      }
      conn_fds_.insert(fd);
      conn_threads_.emplace_back([this, fd] {
        conn_loop(fd);
        std::lock_guard<std::mutex> g2(conn_mu_);
        conn_fds_.erase(fd);
        ::close(fd);
      });
    }
  }

  void conn_loop(int fd) {
    std::string head, payload;
    while (running_ && recv_msg(fd, &head, &payload)) {
      std::string rhead, rpayload;
      handle(head, payload, &rhead, &rpayload);
      if (!send_msg(fd, rhead, rpayload)) break;
    }
  }

  void handle(const std::string& head, const std::string& payload,
              std::string* rhead, std::string* rpayload) {
    std::map<std::string, JsonField> f;
    if (!parse_object(head, &f) || !f.count("op")) {
      *rhead = "{\"ok\": false, \"error\": \"bad message\"}";
      return;
    }
    std::string op = to_string_field(f["op"]);
    if (op == "ping") {
      *rhead = "{\"ok\": true, \"native\": true, \"job\": " +
               json_escape(job_) + ", \"task\": " + std::to_string(task_) +
               "}";
      return;
    }
    if (op == "shutdown") {
      {
        std::lock_guard<std::mutex> g(mu_);
        shutdown_flag_ = true;
      }
      cv_.notify_all();
      *rhead = "{\"ok\": true}";
      return;
    }
    if (is_master_) {
      if (op == "register" || op == "heartbeat") {
        std::lock_guard<std::mutex> g(mu_);
        members_[to_string_field(f["job"]) + ":" +
                 std::to_string((long)to_number(f["task"]))] = now_s();
        cv_.notify_all();
        *rhead = "{\"ok\": true}";
        return;
      }
      if (op == "members") {
        std::lock_guard<std::mutex> g(mu_);
        std::string out = "{\"ok\": true, \"members\": {";
        bool first = true;
        for (auto& kv : members_) {
          if (!first) out += ", ";
          first = false;
          out += json_escape(kv.first) + ": " + std::to_string(kv.second);
        }
        out += "}}";
        *rhead = out;
        return;
      }
      if (op == "put") {
        std::lock_guard<std::mutex> g(mu_);
        kv_[to_string_field(f["key"])] =
            f.count("value") ? f["value"].raw : "null";
        cv_.notify_all();
        *rhead = "{\"ok\": true}";
        return;
      }
      if (op == "get") {
        std::string key = to_string_field(f["key"]);
        bool blocking = f.count("blocking") && f["blocking"].raw == "true";
        double timeout = f.count("timeout") ? to_number(f["timeout"]) : 0.0;
        std::unique_lock<std::mutex> lk(mu_);
        if (blocking) {
          cv_.wait_for(lk, std::chrono::duration<double>(
                               timeout > 0 ? timeout : 120.0),
                       [&] { return kv_.count(key) || shutdown_flag_; });
        }
        if (kv_.count(key)) {
          *rhead = "{\"ok\": true, \"found\": true, \"value\": " + kv_[key] +
                   "}";
        } else {
          *rhead = "{\"ok\": true, \"found\": false}";
        }
        return;
      }
      if (op == "incr") {
        std::lock_guard<std::mutex> g(mu_);
        long amount = f.count("amount") ? (long)to_number(f["amount"]) : 1;
        long v = (counters_[to_string_field(f["key"])] += amount);
        *rhead = "{\"ok\": true, \"value\": " + std::to_string(v) + "}";
        return;
      }
      if (op == "read_counter") {
        std::lock_guard<std::mutex> g(mu_);
        long v = counters_.count(to_string_field(f["key"]))
                     ? counters_[to_string_field(f["key"])]
                     : 0;
        *rhead = "{\"ok\": true, \"value\": " + std::to_string(v) + "}";
        return;
      }
      if (op == "put_counter") {
        std::lock_guard<std::mutex> g(mu_);
        counters_[to_string_field(f["key"])] = (long)to_number(f["value"]);
        *rhead = "{\"ok\": true}";
        return;
      }
      if (op == "barrier") {
        handle_barrier(f, rhead);
        return;
      }
    }
    // unknown / service op -> Python fallback (async PS shard etc.)
    if (has_fallback_) {
      py::gil_scoped_acquire gil;
      try {
        py::tuple r = fallback_(py::bytes(head), py::bytes(payload));
        *rhead = py::cast<std::string>(r[0]);
        *rpayload = py::cast<std::string>(r[1]);
      } catch (const std::exception& e) {
        *rhead = std::string("{\"ok\": false, \"error\": ") +
                 json_escape(e.what()) + "}";
      }
      return;
    }
    *rhead = "{\"ok\": false, \"error\": \"unknown op " + op + "\"}";
  }

  void handle_barrier(std::map<std::string, JsonField>& f,
                      std::string* rhead) {
    std::string name = to_string_field(f["name"]);
    std::string who = to_string_field(f["who"]);
    long count = (long)to_number(f["count"]);
    double timeout = f.count("timeout") ? to_number(f["timeout"]) : 300.0;
    std::unique_lock<std::mutex> lk(mu_);
    auto& b = barriers_[name];
    long gen = b.gen;
    b.arrived.insert(who);
    if ((long)b.arrived.size() >= count) {
      b.gen++;
      b.arrived.clear();
      cv_.notify_all();
      *rhead = "{\"ok\": true}";
      return;
    }
    bool ok = cv_.wait_for(lk, std::chrono::duration<double>(timeout), [&] {
      return barriers_[name].gen != gen || shutdown_flag_;
    });
    *rhead = ok ? "{\"ok\": true}"
                : "{\"ok\": false, \"error\": \"barrier timeout\"}";
  }

  struct Barrier {
    std::set<std::string> arrived;
    long gen = 0;
  };

  int port_;
  bool is_master_;
  std::string job_;
  long task_ = 0;
  int listen_fd_ = -1;
  std::atomic<bool> running_{false};
  std::thread accept_thread_;
  std::mutex conn_mu_;
  std::vector<std::thread> conn_threads_;
  std::set<int> conn_fds_;

  std::mutex mu_;
  std::condition_variable cv_;
  bool shutdown_flag_ = false;
  std::map<std::string, std::string> kv_;  // key -> raw JSON value
  std::map<std::string, long> counters_;
  std::map<std::string, Barrier> barriers_;
  std::map<std::string, double> members_;

  std::mutex py_mu_;
  py::function fallback_;
  std::atomic<bool> has_fallback_{false};
};

}  // namespace

PYBIND11_MODULE(_ccp, m) {
  m.doc() = "rocps native TCP control plane (C1)";
  // Debug aid: ROCPS_TERMINATE_BT=1 prints a native backtrace when
  // std::terminate fires (e.g. a joinable std::thread destroyed), so
  // exit-race aborts can be located without gdb.
  if (const char* bt = std::getenv("ROCPS_TERMINATE_BT"); bt && bt[0] == '1') {
    std::set_terminate([] {
      void* frames[64];
      int n = backtrace(frames, 64);
      const char msg[] = "\n[rocps] std::terminate backtrace:\n";
      (void)!write(2, msg, sizeof(msg) - 1);
      backtrace_symbols_fd(frames, n, 2);
      std::abort();
    });
  }
  py::class_<ControlPlaneServer>(m, "ControlPlaneServer")
      .def(py::init<int, bool, const std::string&, long>(), py::arg("port"),
           py::arg("is_master"), py::arg("job") = "", py::arg("task") = 0)
      .def("start", &ControlPlaneServer::start, py::arg("bind_host") = "",
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &ControlPlaneServer::stop,
           py::call_guard<py::gil_scoped_release>())
      .def("set_fallback", &ControlPlaneServer::set_fallback)
      .def("shutdown_requested", &ControlPlaneServer::shutdown_requested,
           py::call_guard<py::gil_scoped_release>())
      .def("wait_shutdown", &ControlPlaneServer::wait_shutdown,
           py::arg("timeout") = 0.0,
           py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("port", &ControlPlaneServer::port);
}
