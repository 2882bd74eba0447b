// Sanitizer stress driver for the native control plane (SURVEY §5
// race-detection CI): compiles csrc/control/control_plane.cc into a
// standalone binary (no Python interpreter; the pybind fallback path
// is never armed) and hammers one master server from many client
// threads with the real wire protocol — register/heartbeat/put/get/
// incr/barrier — so TSan sees every lock in the kv/membership/barrier
// tables under contention and ASan checks the framing buffers.
//
// Built and run by tools/sanitize/run.sh (ASan and TSan variants);
// tests/test_sanitizers.py wires it into the CPU gate.

#include "../../csrc/control/control_plane.cc"  // the unit under test

#include <cstring>
#include <sstream>
#include <vector>

namespace {

int connect_to(int port) {
  int fd = ::socket(AF_INET, SOCK_STREAM, 0);
  if (fd < 0) return -1;
  sockaddr_in a{};
  a.sin_family = AF_INET;
  a.sin_port = htons((uint16_t)port);
  a.sin_addr.s_addr = inet_addr("127.0.0.1");
  if (::connect(fd, (sockaddr*)&a, sizeof(a)) != 0) {
    ::close(fd);
    return -1;
  }
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
  return fd;
}

// send_all / recv_all come from the included control_plane.cc

// wire: u64le total(=8+jlen+plen) | u64le jlen | json | payload
bool call(int fd, const std::string& json, std::string* reply) {
  uint64_t jlen = json.size(), total = 8 + jlen;
  char hdr[16];
  memcpy(hdr, &total, 8);
  memcpy(hdr + 8, &jlen, 8);
  if (!send_all(fd, hdr, 16) || !send_all(fd, json.data(), jlen))
    return false;
  char rh[16];
  if (!recv_all(fd, rh, 16)) return false;
  uint64_t rtotal, rjlen;
  memcpy(&rtotal, rh, 8);
  memcpy(&rjlen, rh + 8, 8);
  if (rtotal < 8 + rjlen || rtotal > (64u << 20)) return false;
  std::string body(rtotal - 8, '\0');
  if (!recv_all(fd, body.data(), body.size())) return false;
  if (reply) *reply = body.substr(0, rjlen);
  return true;
}

std::atomic<int> failures{0};

void worker(int port, int id, int nthreads, int ops) {
  int fd = connect_to(port);
  if (fd < 0) {
    failures++;
    return;
  }
  std::string r;
  std::ostringstream reg;
  reg << "{\"op\":\"register\",\"job\":\"worker\",\"task\":" << id << "}";
  if (!call(fd, reg.str(), &r)) failures++;
  for (int i = 0; i < ops; i++) {
    std::ostringstream hb, put, get, inc;
    hb << "{\"op\":\"heartbeat\",\"job\":\"worker\",\"task\":" << id << "}";
    put << "{\"op\":\"put\",\"key\":\"k" << (i % 7) << "\",\"value\":\"v"
        << id << "_" << i << "\"}";
    get << "{\"op\":\"get\",\"key\":\"k" << (i % 7)
        << "\",\"blocking\":false}";
    inc << "{\"op\":\"incr\",\"key\":\"ctr\",\"amount\":1}";
    if (!call(fd, hb.str(), &r) || !call(fd, put.str(), &r) ||
        !call(fd, get.str(), &r) || !call(fd, inc.str(), &r)) {
      failures++;
      break;
    }
    if (i % 10 == 5) {
      std::ostringstream bar;
      bar << "{\"op\":\"barrier\",\"name\":\"b" << (i / 10)
          << "\",\"who\":\"w" << id << "\",\"count\":" << nthreads
          << ",\"timeout\":30}";
      if (!call(fd, bar.str(), &r) ||
          r.find("\"ok\": true") == std::string::npos) {
        failures++;
        break;
      }
    }
  }
  ::close(fd);
}

}  // namespace

int main() {
  ControlPlaneServer server(0, /*is_master=*/true, "ps", 0);
  int port = server.start("127.0.0.1");
  const int T = 8, OPS = 40;
  std::vector<std::thread> ts;
  for (int i = 0; i < T; i++)
    ts.emplace_back(worker, port, i, T, OPS);
  for (auto& t : ts) t.join();
  // final consistency: the counter saw every increment exactly once
  int fd = connect_to(port);
  std::string r;
  if (fd < 0 || !call(fd, "{\"op\":\"read_counter\",\"key\":\"ctr\"}", &r)) {
    fprintf(stderr, "final read failed\n");
    return 2;
  }
  ::close(fd);
  char want[64];
  snprintf(want, sizeof(want), "\"value\": %d", T * OPS);
  if (r.find(want) == std::string::npos) {
    fprintf(stderr, "counter mismatch: %s (want %s)\n", r.c_str(), want);
    return 3;
  }
  server.stop();
  if (failures.load()) {
    fprintf(stderr, "%d client failures\n", failures.load());
    return 4;
  }
  printf("control_stress OK: %d threads x %d ops, counter=%d\n", T, OPS,
         T * OPS);
  return 0;
}
