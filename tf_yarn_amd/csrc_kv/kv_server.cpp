// Native control-plane KV server for tf_yarn_amd (the skein
// ApplicationMaster KV replacement).
//
// The reference routes every coordination primitive (barriers, master
// election, cluster-spec exchange, exception propagation, lifecycle
// timing) through the skein application KV store — a Java/gRPC service.
// This is the MI355X-node-native equivalent: a C++ TCP server with
// blocking WAIT and prefix WATCH semantics, exposed to Python via
// pybind11 (tf_yarn_amd.kv picks it over the pure-Python fallback).
//
// Wire protocol (language-neutral, little-endian):
//   request:  [u32 frame_len][u8 op][u16 key_len][key]
//             [u64 payload_len][payload]
//   ops: 1=PUT 2=GET 3=WAIT(payload=8B timeout_ms, 0=inf) 4=DEL
//        5=LIST(prefix=key) 6=WATCH(prefix=key) 7=ADD(payload=8B i64)
//        8=CAS(payload = u64 exp_len | expected | desired)
//   response: [u32 frame_len][u8 status][u64 payload_len][payload]
//   status: 0=ok-with-value 1=ok-empty 2=not-found 3=timeout 4=error
//   LIST payload: repeated [u16 klen][key][u64 vlen][value]
//   WATCH: one ok-empty response, then a stream of
//          [u32 len][u8 10][u16 klen][key][u64 vlen][value] events;
//          [u32 len][u8 11] on server close.

#include <arpa/inet.h>
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
#include <string>
#include <thread>
#include <vector>

#include <pybind11/pybind11.h>

namespace py = pybind11;

namespace {

struct Watcher {
  std::string prefix;
  int fd;
  std::mutex* send_mu;
};

class KvServer {
 public:
  KvServer() : running_(false), listen_fd_(-1), port_(0) {}

  ~KvServer() { stop(); }

  int start(const std::string& host, int port) {
    listen_fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
    if (listen_fd_ < 0) throw std::runtime_error("socket() failed");
    int one = 1;
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons(static_cast<uint16_t>(port));
    inet_pton(AF_INET, host.c_str(), &addr.sin_addr);
    if (bind(listen_fd_, reinterpret_cast<sockaddr*>(&addr),
             sizeof(addr)) != 0)
      throw std::runtime_error("bind() failed");
    if (listen(listen_fd_, 128) != 0)
      throw std::runtime_error("listen() failed");
    socklen_t len = sizeof(addr);
    getsockname(listen_fd_, reinterpret_cast<sockaddr*>(&addr), &len);
    port_ = ntohs(addr.sin_port);
    running_ = true;
    accept_thread_ = std::thread([this] { accept_loop(); });
    return port_;
  }

  int port() const { return port_; }

  void stop() {
    bool expected = true;
    if (!running_.compare_exchange_strong(expected, false)) return;
    {
      std::lock_guard<std::mutex> lk(mu_);
      cv_.notify_all();
      for (auto& w : watchers_) {
        std::lock_guard<std::mutex> slk(*w.send_mu);
        // close event: [u32 len=1][u8 11]
        uint8_t buf[5];
        uint32_t l = 1;
        memcpy(buf, &l, 4);
        buf[4] = 11;
        (void)!::send(w.fd, buf, 5, MSG_NOSIGNAL);
        ::shutdown(w.fd, SHUT_RDWR);
      }
      watchers_.clear();
    }
    ::shutdown(listen_fd_, SHUT_RDWR);
    ::close(listen_fd_);
    if (accept_thread_.joinable()) accept_thread_.join();
    std::lock_guard<std::mutex> lk(threads_mu_);
    // Unblock every connection thread still parked in recv().
    for (int fd : conn_fds_) ::shutdown(fd, SHUT_RDWR);
    for (auto& t : conn_threads_)
      if (t.joinable()) t.join();
    conn_threads_.clear();
    for (int fd : conn_fds_) ::close(fd);
    conn_fds_.clear();
  }

  size_t size() {
    std::lock_guard<std::mutex> lk(mu_);
    return data_.size();
  }

 private:
  // -- io helpers ---------------------------------------------------------
  static bool read_exact(int fd, void* buf, size_t n) {
    auto* p = static_cast<uint8_t*>(buf);
    while (n > 0) {
      ssize_t r = ::recv(fd, p, n, 0);
      if (r <= 0) return false;
      p += r;
      n -= static_cast<size_t>(r);
    }
    return true;
  }

  static bool send_all(int fd, const void* buf, size_t n) {
    auto* p = static_cast<const uint8_t*>(buf);
    while (n > 0) {
      ssize_t r = ::send(fd, p, n, MSG_NOSIGNAL);
      if (r <= 0) return false;
      p += r;
      n -= static_cast<size_t>(r);
    }
    return true;
  }

  static bool send_response(int fd, std::mutex* send_mu, uint8_t status,
                            const std::string& payload) {
    std::vector<uint8_t> out(4 + 1 + 8 + payload.size());
    uint32_t frame = static_cast<uint32_t>(1 + 8 + payload.size());
    uint64_t plen = payload.size();
    memcpy(out.data(), &frame, 4);
    out[4] = status;
    memcpy(out.data() + 5, &plen, 8);
    memcpy(out.data() + 13, payload.data(), payload.size());
    if (send_mu) {
      std::lock_guard<std::mutex> lk(*send_mu);
      return send_all(fd, out.data(), out.size());
    }
    return send_all(fd, out.data(), out.size());
  }

  // -- server loops -------------------------------------------------------
  void accept_loop() {
    while (running_) {
      int fd = ::accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) {
        if (!running_) return;
        continue;
      }
      int one = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      std::lock_guard<std::mutex> lk(threads_mu_);
      conn_fds_.push_back(fd);
      conn_threads_.emplace_back([this, fd] { serve(fd); });
    }
  }

  void notify_watchers(const std::string& key, const std::string& value) {
    std::vector<Watcher> snapshot;
    {
      std::lock_guard<std::mutex> lk(mu_);
      snapshot = watchers_;
    }
    for (auto& w : snapshot) {
      if (key.compare(0, w.prefix.size(), w.prefix) == 0 ||
          w.prefix.empty()) {
        send_event(w, key, value);
      }
    }
  }

  static bool send_event(const Watcher& w, const std::string& key,
                         const std::string& value) {
    std::vector<uint8_t> out(4 + 1 + 2 + key.size() + 8 + value.size());
    uint32_t frame = static_cast<uint32_t>(
        1 + 2 + key.size() + 8 + value.size());
    uint16_t klen = static_cast<uint16_t>(key.size());
    uint64_t vlen = value.size();
    size_t o = 0;
    memcpy(out.data() + o, &frame, 4); o += 4;
    out[o++] = 10;
    memcpy(out.data() + o, &klen, 2); o += 2;
    memcpy(out.data() + o, key.data(), key.size()); o += key.size();
    memcpy(out.data() + o, &vlen, 8); o += 8;
    memcpy(out.data() + o, value.data(), value.size());
    std::lock_guard<std::mutex> lk(*w.send_mu);
    return send_all(w.fd, out.data(), out.size());
  }

  void serve(int fd) {
    auto send_mu = std::make_shared<std::mutex>();
    bool is_watcher = false;
    while (running_) {
      uint32_t frame;
      if (!read_exact(fd, &frame, 4)) break;
      if (frame > (256u << 20)) break;  // corrupt length guard
      std::vector<uint8_t> body(frame);
      if (!read_exact(fd, body.data(), frame)) break;
      if (frame < 3) break;
      uint8_t op = body[0];
      uint16_t klen;
      memcpy(&klen, body.data() + 1, 2);
      if (3 + klen + 8 > frame) break;
      std::string key(reinterpret_cast<char*>(body.data() + 3), klen);
      uint64_t plen;
      memcpy(&plen, body.data() + 3 + klen, 8);
      if (3 + static_cast<uint64_t>(klen) + 8 + plen > frame) break;
      std::string payload(
          reinterpret_cast<char*>(body.data() + 3 + klen + 8), plen);

      switch (op) {
        case 1: {  // PUT
          {
            std::lock_guard<std::mutex> lk(mu_);
            data_[key] = payload;
            cv_.notify_all();
          }
          notify_watchers(key, payload);
          send_response(fd, send_mu.get(), 1, "");
          break;
        }
        case 2: {  // GET
          std::unique_lock<std::mutex> lk(mu_);
          auto it = data_.find(key);
          if (it == data_.end()) {
            lk.unlock();
            send_response(fd, send_mu.get(), 2, "");
          } else {
            std::string v = it->second;
            lk.unlock();
            send_response(fd, send_mu.get(), 0, v);
          }
          break;
        }
        case 3: {  // WAIT
          uint64_t timeout_ms = 0;
          if (payload.size() >= 8) memcpy(&timeout_ms, payload.data(), 8);
          std::unique_lock<std::mutex> lk(mu_);
          auto pred = [&] {
            return data_.count(key) > 0 || !running_;
          };
          bool ok;
          if (timeout_ms == 0) {
            cv_.wait(lk, pred);
            ok = data_.count(key) > 0;
          } else {
            ok = cv_.wait_for(lk, std::chrono::milliseconds(timeout_ms),
                              pred) && data_.count(key) > 0;
          }
          if (ok) {
            std::string v = data_[key];
            lk.unlock();
            send_response(fd, send_mu.get(), 0, v);
          } else {
            lk.unlock();
            send_response(fd, send_mu.get(), 3, "");
          }
          break;
        }
        case 4: {  // DEL
          std::lock_guard<std::mutex> lk(mu_);
          data_.erase(key);
          send_response(fd, send_mu.get(), 1, "");
          break;
        }
        case 5: {  // LIST
          std::string out;
          {
            std::lock_guard<std::mutex> lk(mu_);
            for (auto& kv : data_) {
              if (kv.first.compare(0, key.size(), key) != 0) continue;
              uint16_t kl = static_cast<uint16_t>(kv.first.size());
              uint64_t vl = kv.second.size();
              out.append(reinterpret_cast<char*>(&kl), 2);
              out.append(kv.first);
              out.append(reinterpret_cast<char*>(&vl), 8);
              out.append(kv.second);
            }
          }
          send_response(fd, send_mu.get(), 0, out);
          break;
        }
        case 6: {  // WATCH: register, replay existing, stream
          std::vector<std::pair<std::string, std::string>> existing;
          {
            std::lock_guard<std::mutex> lk(mu_);
            for (auto& kv : data_)
              if (key.empty() ||
                  kv.first.compare(0, key.size(), key) == 0)
                existing.push_back(kv);
            watcher_mus_.push_back(send_mu);
            watchers_.push_back({key, fd, send_mu.get()});
          }
          send_response(fd, send_mu.get(), 1, "");
          Watcher self{key, fd, send_mu.get()};
          for (auto& kv : existing) send_event(self, kv.first, kv.second);
          is_watcher = true;
          break;
        }
        case 7: {  // ADD
          int64_t amount = 0;
          if (payload.size() >= 8) memcpy(&amount, payload.data(), 8);
          int64_t result;
          std::string sval;
          {
            std::lock_guard<std::mutex> lk(mu_);
            int64_t cur = 0;
            auto it = data_.find(key);
            if (it != data_.end() && !it->second.empty())
              cur = std::stoll(it->second);
            result = cur + amount;
            sval = std::to_string(result);
            data_[key] = sval;
            cv_.notify_all();
          }
          notify_watchers(key, sval);
          send_response(fd, send_mu.get(), 0, std::to_string(result));
          break;
        }
        case 8: {  // CAS: payload = u64 exp_len | expected | desired
          if (payload.size() < 8) {
            send_response(fd, send_mu.get(), 4, "bad cas");
            break;
          }
          uint64_t elen;
          memcpy(&elen, payload.data(), 8);
          std::string expected = payload.substr(8, elen);
          std::string desired = payload.substr(8 + elen);
          std::string result;
          bool changed = false;
          {
            std::lock_guard<std::mutex> lk(mu_);
            auto it = data_.find(key);
            if ((it == data_.end() && expected.empty()) ||
                (it != data_.end() && it->second == expected)) {
              data_[key] = desired;
              result = desired;
              changed = true;
              cv_.notify_all();
            } else {
              result = it != data_.end() ? it->second : expected;
            }
          }
          if (changed) notify_watchers(key, desired);
          send_response(fd, send_mu.get(), 0, result);
          break;
        }
        default:
          send_response(fd, send_mu.get(), 4, "unknown op");
      }
      if (is_watcher) return;  // fd now owned by the watcher stream
    }
    // fd stays registered in conn_fds_; stop() closes it exactly once.
    if (!is_watcher) ::shutdown(fd, SHUT_RDWR);
  }

  std::atomic<bool> running_;
  int listen_fd_;
  int port_;
  std::thread accept_thread_;
  std::mutex threads_mu_;
  std::vector<std::thread> conn_threads_;
  std::vector<int> conn_fds_;
  std::mutex mu_;
  std::condition_variable cv_;
  std::map<std::string, std::string> data_;
  std::vector<Watcher> watchers_;
  std::vector<std::shared_ptr<std::mutex>> watcher_mus_;
};

}  // namespace

PYBIND11_MODULE(_kv_native, m) {
  m.doc() = "Native control-plane KV server (C++)";
  py::class_<KvServer>(m, "KvServer")
      .def(py::init<>())
      .def("start", &KvServer::start, py::arg("host") = "127.0.0.1",
           py::arg("port") = 0,
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &KvServer::stop,
           py::call_guard<py::gil_scoped_release>())
      .def("port", &KvServer::port)
      .def("size", &KvServer::size,
           py::call_guard<py::gil_scoped_release>());
}
