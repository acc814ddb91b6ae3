// nerrfd — native collector daemon for the nerrf-amd tracker.
//
// Capture sources:
//   * --replay <trace.jsonl|.csv>: replay a recorded trace (always available;
//     the path exercised by tests in this repo),
//   * eBPF ring buffer (tracker/bpf/nerrf_tracepoints.c) when built with
//     -DNERRF_HAVE_LIBBPF on a host with libbpf — the deployment mode.
//
// Transport: length-prefixed (u32 LE) nerrf.trace.EventBatch frames over TCP
// with per-client bounded queues and drop-on-slow-client semantics (the
// upstream tracker's operational contract: a stalled consumer never blocks
// capture).  nerrf_amd/serve/daemon_bridge.py re-exposes the stream as the
// gRPC Tracker/StreamEvents contract for wire-level compatibility (this
// image ships no grpc++; the hot path — parse, batch, serialize — is native).
//
// Build: make -C tracker daemon     (g++ -O2, no external deps)
#include <arpa/inet.h>
#include <netinet/in.h>
#include <sys/socket.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <fstream>
#include <iostream>
#include <mutex>
#include <sstream>
#include <thread>
#include <vector>

#include "wire_codec.h"

#ifdef NERRF_HAVE_LIBBPF
#include <bpf/libbpf.h>
#include "../bpf/event_abi.h"
#endif

namespace {

using nerrf::wire::Event;

struct Client {
  int fd;
  std::deque<std::string> queue;  // bounded; drop-on-full
  size_t dropped = 0;
};

class Broadcaster {
 public:
  explicit Broadcaster(size_t max_queue = 100) : max_queue_(max_queue) {}

  void add_client(int fd) {
    std::lock_guard<std::mutex> lk(mu_);
    clients_.push_back({fd});
  }

  void broadcast(std::string frame) {
    std::lock_guard<std::mutex> lk(mu_);
    for (auto& c : clients_) {
      if (c.queue.size() >= max_queue_) {
        ++c.dropped;  // slow client: drop, never block capture
        continue;
      }
      c.queue.push_back(frame);
    }
    cv_.notify_all();
  }

  // Pump queued frames to sockets; removes dead clients.
  void run(std::atomic<bool>& stop) {
    while (!stop.load()) {
      std::unique_lock<std::mutex> lk(mu_);
      cv_.wait_for(lk, std::chrono::milliseconds(100));
      for (auto it = clients_.begin(); it != clients_.end();) {
        bool dead = false;
        while (!it->queue.empty()) {
          const std::string& frame = it->queue.front();
          uint32_t len = htonl(uint32_t(frame.size()));
          if (!send_all(it->fd, &len, 4) ||
              !send_all(it->fd, frame.data(), frame.size())) {
            dead = true;
            break;
          }
          it->queue.pop_front();
        }
        if (dead) {
          close(it->fd);
          it = clients_.erase(it);
        } else {
          ++it;
        }
      }
    }
  }

  size_t n_clients() {
    std::lock_guard<std::mutex> lk(mu_);
    return clients_.size();
  }

 private:
  static bool send_all(int fd, const void* buf, size_t len) {
    const char* p = static_cast<const char*>(buf);
    while (len) {
      ssize_t n = send(fd, p, len, MSG_NOSIGNAL);
      if (n <= 0) return false;
      p += n;
      len -= size_t(n);
    }
    return true;
  }

  std::mutex mu_;
  std::condition_variable cv_;
  std::vector<Client> clients_;
  size_t max_queue_;
};

// ---- replay source: jsonl ({"timestamp","event","path","size","pid"}) -----

std::string json_str(const std::string& line, const std::string& key) {
  auto kpos = line.find("\"" + key + "\"");
  if (kpos == std::string::npos) return "";
  auto colon = line.find(':', kpos);
  if (colon == std::string::npos) return "";
  auto q1 = line.find('"', colon + 1);
  auto comma = line.find_first_of(",}", colon + 1);
  if (q1 != std::string::npos && (comma == std::string::npos || q1 < comma)) {
    auto q2 = line.find('"', q1 + 1);
    return line.substr(q1 + 1, q2 - q1 - 1);
  }
  // numeric
  auto start = line.find_first_not_of(" \t", colon + 1);
  auto end = line.find_first_of(",}", start);
  return line.substr(start, end - start);
}

std::vector<Event> load_jsonl(const std::string& path) {
  std::ifstream in(path);
  std::vector<Event> events;
  std::string line;
  while (std::getline(in, line)) {
    if (line.empty()) continue;
    Event ev;
    double ts = atof(json_str(line, "timestamp").c_str());
    ev.ts_sec = int64_t(ts);
    ev.ts_nsec = int32_t((ts - double(ev.ts_sec)) * 1e9);
    ev.syscall = json_str(line, "event");
    if (ev.syscall == "open" || ev.syscall == "create") ev.syscall = "openat";
    if (ev.syscall == "delete") ev.syscall = "unlink";
    ev.path = json_str(line, "path");
    ev.new_path = json_str(line, "new_path");
    ev.bytes = uint64_t(atoll(json_str(line, "size").c_str()));
    ev.pid = uint32_t(atoi(json_str(line, "pid").c_str()));
    ev.tid = ev.pid;
    events.push_back(std::move(ev));
  }
  return events;
}

#ifdef NERRF_HAVE_LIBBPF
// ---- live source: eBPF ring buffer ----------------------------------------
// capture timestamps are CLOCK_MONOTONIC (bpf_ktime_get_ns); convert to wall
// clock with the boot-time offset computed once at startup (the upstream
// tracker's contract: absolute UTC timestamps on the wire)
static uint64_t mono_to_wall_offset_ns() {
  timespec mono{}, wall{};
  clock_gettime(CLOCK_MONOTONIC, &mono);
  clock_gettime(CLOCK_REALTIME, &wall);
  uint64_t m = uint64_t(mono.tv_sec) * 1000000000ull + mono.tv_nsec;
  uint64_t w = uint64_t(wall.tv_sec) * 1000000000ull + wall.tv_nsec;
  return w - m;
}

int handle_ringbuf_event(void* ctx, void* data, size_t size) {
  if (size < sizeof(nerrf_event)) return 0;
  static const uint64_t boot_offset = mono_to_wall_offset_ns();
  auto* raw = static_cast<const nerrf_event*>(data);
  auto* out = static_cast<std::vector<Event>*>(ctx);
  Event ev;
  const uint64_t wall_ns = raw->ts_ns + boot_offset;
  ev.ts_sec = int64_t(wall_ns / 1000000000ull);
  ev.ts_nsec = int32_t(wall_ns % 1000000000ull);
  ev.pid = raw->pid;
  ev.tid = raw->tid;
  ev.comm = raw->comm;
  static const char* names[] = {"unknown", "openat", "write", "rename",
                                "read",    "unlink", "chmod"};
  ev.syscall = names[raw->syscall_id >= 0 && raw->syscall_id <= 6
                         ? raw->syscall_id : 0];
  ev.path = raw->path;
  ev.new_path = raw->new_path;
  ev.flags = uint32_t(raw->flags);
  ev.bytes = raw->bytes;
  out->push_back(std::move(ev));
  return 0;
}
#endif

}  // namespace

int main(int argc, char** argv) {
  std::string replay_path;
  int port = 50052;
  int batch_size = 64;
  double rate = 0.0;  // 0 = as fast as possible
  bool wait_client = true;
  bool once = false;
  // capture-side path-prefix filtering (upstream M2 plan,
  // overview.mdx:262-269: cut event volume before it leaves the node) —
  // the same prefix set the eBPF program's path_filter map enforces
  // in-kernel on a live host applies here to replay/live streams
  std::vector<std::string> filter_prefixes;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    if (a == "--replay" && i + 1 < argc) replay_path = argv[++i];
    else if (a == "--port" && i + 1 < argc) port = atoi(argv[++i]);
    else if (a == "--batch" && i + 1 < argc) batch_size = atoi(argv[++i]);
    else if (a == "--rate" && i + 1 < argc) rate = atof(argv[++i]);
    else if (a == "--filter-prefix" && i + 1 < argc)
      filter_prefixes.push_back(argv[++i]);
    else if (a == "--no-wait") wait_client = false;
    else if (a == "--once") once = true;  // exit after one replay pass
    else {
      std::cerr << "usage: nerrfd [--replay trace.jsonl] [--port P] "
                   "[--batch N] [--rate X] [--filter-prefix P]... "
                   "[--no-wait] [--once]\n";
      return 2;
    }
  }
  auto passes_filter = [&](const Event& ev) {
    if (filter_prefixes.empty()) return true;
    // pathless events (e.g. unresolved writes) always pass: dropping them
    // would hide byte-volume signal the detector consumes
    if (ev.path.empty() && ev.new_path.empty()) return true;
    for (const auto& p : filter_prefixes) {
      if (ev.path.compare(0, p.size(), p) == 0) return true;
      if (!ev.new_path.empty() && ev.new_path.compare(0, p.size(), p) == 0)
        return true;
    }
    return false;
  };

  int srv = socket(AF_INET, SOCK_STREAM, 0);
  int one = 1;
  setsockopt(srv, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
  sockaddr_in addr{};
  addr.sin_family = AF_INET;
  addr.sin_addr.s_addr = htonl(INADDR_LOOPBACK);
  addr.sin_port = htons(uint16_t(port));
  if (bind(srv, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) != 0) {
    perror("bind");
    return 1;
  }
  listen(srv, 16);
  socklen_t alen = sizeof(addr);
  getsockname(srv, reinterpret_cast<sockaddr*>(&addr), &alen);
  std::cout << "nerrfd listening on 127.0.0.1:" << ntohs(addr.sin_port)
            << std::endl;

  Broadcaster bc;
  std::atomic<bool> stop{false};
  std::thread pump([&] { bc.run(stop); });
  std::thread acceptor([&] {
    while (!stop.load()) {
      int fd = accept(srv, nullptr, nullptr);
      if (fd >= 0) bc.add_client(fd);
      else if (stop.load()) break;
    }
  });

  if (!replay_path.empty()) {
    auto events = load_jsonl(replay_path);
    std::cerr << "replaying " << events.size() << " events from "
              << replay_path << std::endl;
    if (wait_client) {
      while (!stop.load() && bc.n_clients() == 0)
        std::this_thread::sleep_for(std::chrono::milliseconds(10));
    }
    std::vector<Event> batch;
    double t_prev = -1;
    for (auto& ev : events) {
      if (rate > 0 && t_prev >= 0) {
        double dt = (ev.timestamp() - t_prev) / rate;
        if (dt > 0)
          std::this_thread::sleep_for(std::chrono::duration<double>(
              std::min(dt, 0.5)));
      }
      t_prev = ev.timestamp();
      if (!passes_filter(ev)) continue;
      batch.push_back(std::move(ev));
      if (int(batch.size()) >= batch_size) {
        bc.broadcast(nerrf::wire::encode_event_batch(batch));
        batch.clear();
      }
    }
    if (!batch.empty()) bc.broadcast(nerrf::wire::encode_event_batch(batch));
    // give the pump a moment to flush, then exit if --once
    std::this_thread::sleep_for(std::chrono::milliseconds(300));
    if (once) {
      stop.store(true);
      shutdown(srv, SHUT_RDWR);
      close(srv);
      pump.join();
      acceptor.join();
      return 0;
    }
  }
#ifdef NERRF_HAVE_LIBBPF
  else {
    // live mode: load tracker/bpf/nerrf_tracepoints.o, attach, poll ringbuf
    struct bpf_object* obj = bpf_object__open("nerrf_tracepoints.o");
    if (!obj || bpf_object__load(obj)) {
      std::cerr << "failed to load nerrf_tracepoints.o" << std::endl;
      return 1;
    }
    bpf_program* prog;
    bpf_object__for_each_program(prog, obj) {
      if (!bpf_program__attach(prog)) {
        std::cerr << "attach failed: " << bpf_program__section_name(prog)
                  << std::endl;
      }
    }
    int map_fd = bpf_object__find_map_fd_by_name(obj, "events");
    std::vector<Event> pending;
    ring_buffer* rb =
        ring_buffer__new(map_fd, handle_ringbuf_event, &pending, nullptr);
    while (!stop.load()) {
      ring_buffer__poll(rb, 100);
      if (int(pending.size()) >= batch_size) {
        bc.broadcast(nerrf::wire::encode_event_batch(pending));
        pending.clear();
      }
    }
  }
#else
  else {
    std::cerr << "built without libbpf: live capture unavailable; "
                 "use --replay" << std::endl;
  }
#endif

  while (!stop.load()) std::this_thread::sleep_for(std::chrono::seconds(1));
  pump.join();
  acceptor.join();
  return 0;
}
