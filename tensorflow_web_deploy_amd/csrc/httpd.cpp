// httpd.cpp — C++ epoll HTTP/1.1 frontend for the serving fast path.
//
// Why this exists (measured, BASELINE.md "serving analysis"): with the
// Flask/werkzeug stack every request costs Python-thread time, and the
// batcher thread is GIL-starved behind dozens of request threads — a
// batch whose GPU+decode work is 0.3-3 ms takes 73-91 ms wall in situ.
// This frontend moves accept/parse/batch-window/respond into C++
// threads that never take the GIL; the Python process spends its
// interpreter time exclusively on whole batches (decode pool dispatch,
// HIP engine, JSON), never on individual requests.
//
// Surface (bound as tensorflow_web_deploy_amd._httpd.Frontend):
//   Frontend(host, port, workers, backlog)   port=0 -> ephemeral
//   .port()                         actual bound port
//   .wait_batch(max_batch, max_wait_ms) -> [(id, bytes), ...]
//       blocks (GIL released) until the first request, then closes the
//       window at max_batch or max_wait_ms after the first arrival —
//       same policy as sched/batcher.py DynamicBatcher.
//   .respond_many([(id, status, json_bytes), ...])
//   .stats() / .stop()
//
// GET /healthz, 404s, 413s and queue-full 503s are answered inline in
// C++; only POST /predict bodies cross into Python.

#include <arpa/inet.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/epoll.h>
#include <sys/eventfd.h>
#include <sys/socket.h>
#include <unistd.h>

#include <atomic>
#include <cctype>
#include <chrono>
#include <cstdlib>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <memory>
#include <mutex>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

namespace py = pybind11;

namespace {

constexpr size_t kMaxBody = 32u << 20;  // 413 beyond this
constexpr int kMaxEvents = 128;

struct Conn {
  int fd = -1;
  uint32_t gen = 0;
  int64_t last_ms = 0;  // last read/write activity (idle sweep)
  std::string in;
  std::string out;
  size_t out_off = 0;
  // HTTP/1.1 requires in-order responses; Python completes batches in
  // arbitrary per-request order, so we parse one request per conn at a
  // time (request/response lockstep — what urllib/requests/curl do).
  bool awaiting = false;
  bool keep_alive = true;
  bool closing = false;    // close once out drains
  bool rd_closed = false;  // peer FIN seen; stop polling EPOLLIN
  bool sent_100 = false;
};

struct Req {
  uint64_t id;
  std::string body;
};

struct Resp {
  uint64_t id;
  int status;
  std::string body;
};

const char* status_text(int s) {
  switch (s) {
    case 200: return "OK";
    case 400: return "Bad Request";
    case 404: return "Not Found";
    case 411: return "Length Required";
    case 413: return "Payload Too Large";
    case 503: return "Service Unavailable";
    default: return "Error";
  }
}

std::string http_response(int status, const std::string& body,
                          bool keep_alive) {
  std::string r;
  r.reserve(body.size() + 128);
  r += "HTTP/1.1 ";
  r += std::to_string(status);
  r += ' ';
  r += status_text(status);
  r += "\r\nContent-Type: application/json\r\nContent-Length: ";
  r += std::to_string(body.size());
  r += keep_alive ? "\r\nConnection: keep-alive\r\n\r\n"
                  : "\r\nConnection: close\r\n\r\n";
  r += body;
  return r;
}

bool iequal(const char* a, const char* b, size_t n) {
  for (size_t i = 0; i < n; ++i)
    if (::tolower((unsigned char)a[i]) != ::tolower((unsigned char)b[i]))
      return false;
  return true;
}

class Frontend {
 public:
  Frontend(const std::string& host, int port, int workers, int backlog,
           int idle_timeout_s = 120)
      : nworkers_(workers < 1 ? 1 : (workers > 15 ? 15 : workers)),
        backlog_(backlog), idle_timeout_ms_(idle_timeout_s * 1000LL) {
    listen_fd_ = ::socket(AF_INET, SOCK_STREAM | SOCK_NONBLOCK, 0);
    if (listen_fd_ < 0) throw std::runtime_error("socket() failed");
    int one = 1;
    ::setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons((uint16_t)port);
    if (::inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1)
      addr.sin_addr.s_addr = htonl(INADDR_ANY);
    if (::bind(listen_fd_, (sockaddr*)&addr, sizeof(addr)) != 0) {
      ::close(listen_fd_);
      throw std::runtime_error("bind() failed: " +
                               std::string(strerror(errno)));
    }
    if (::listen(listen_fd_, 1024) != 0) {
      ::close(listen_fd_);
      throw std::runtime_error("listen() failed");
    }
    sockaddr_in got{};
    socklen_t gl = sizeof(got);
    ::getsockname(listen_fd_, (sockaddr*)&got, &gl);
    port_ = ntohs(got.sin_port);

    running_.store(true);
    ws_.reserve(nworkers_);
    for (int i = 0; i < nworkers_; ++i) ws_.emplace_back(new Worker);
    for (int i = 0; i < nworkers_; ++i) {
      Worker& w = *ws_[i];
      w.epfd = ::epoll_create1(EPOLL_CLOEXEC);
      w.evfd = ::eventfd(0, EFD_NONBLOCK | EFD_CLOEXEC);
      epoll_event ev{};
      ev.events = EPOLLIN;
      ev.data.fd = w.evfd;
      ::epoll_ctl(w.epfd, EPOLL_CTL_ADD, w.evfd, &ev);
      if (i == 0) {  // worker 0 owns the acceptor
        epoll_event le{};
        le.events = EPOLLIN;
        le.data.fd = listen_fd_;
        ::epoll_ctl(w.epfd, EPOLL_CTL_ADD, listen_fd_, &le);
      }
      w.th = std::thread([this, i] { worker_loop(i); });
    }
  }

  ~Frontend() { stop(); }

  int port() const { return port_; }

  void stop() {
    bool was = running_.exchange(false);
    if (!was) return;
    qcv_.notify_all();
    for (auto& wp : ws_) ding(*wp);
    for (auto& wp : ws_)
      if (wp->th.joinable()) wp->th.join();
    if (listen_fd_ >= 0) ::close(listen_fd_);
    listen_fd_ = -1;
  }

  // --- Python-facing batch API -------------------------------------
  std::vector<std::pair<uint64_t, py::bytes>> wait_batch(int max_batch,
                                                         double max_wait_ms) {
    std::vector<Req> got;
    {
      py::gil_scoped_release nogil;
      std::unique_lock<std::mutex> lk(qmu_);
      // block for the 1st request, but give the GIL back after ~1 s
      // idle: Python signal handlers (SIGTERM/SIGINT) only run between
      // bytecodes of the main thread, so an unbounded C++ wait here
      // makes the server unkillable while idle
      auto idle_deadline = std::chrono::steady_clock::now() +
                           std::chrono::milliseconds(1000);
      while (running_.load() && reqq_.empty()) {
        if (std::chrono::steady_clock::now() >= idle_deadline) break;
        qcv_.wait_for(lk, std::chrono::milliseconds(100));
      }
      if (!reqq_.empty()) {
        auto deadline = std::chrono::steady_clock::now() +
                        std::chrono::duration<double, std::milli>(max_wait_ms);
        while ((int)reqq_.size() < max_batch && running_.load() &&
               qcv_.wait_until(lk, deadline) != std::cv_status::timeout) {
        }
        int n = (int)reqq_.size();
        if (n > max_batch) n = max_batch;
        got.reserve(n);
        for (int i = 0; i < n; ++i) {
          got.emplace_back(std::move(reqq_.front()));
          reqq_.pop_front();
        }
      }
    }
    std::vector<std::pair<uint64_t, py::bytes>> out;
    out.reserve(got.size());
    for (auto& r : got) out.emplace_back(r.id, py::bytes(r.body));
    return out;
  }

  void respond_many(
      const std::vector<std::tuple<uint64_t, int, py::bytes>>& rs) {
    // copy payloads under the GIL, deliver without it
    std::vector<Resp> byw[16];
    for (auto& t : rs) {
      uint64_t id = std::get<0>(t);
      int w = (int)(id >> 60);
      if (w >= nworkers_) continue;
      byw[w].push_back(
          Resp{id, std::get<1>(t), std::string(std::get<2>(t))});
    }
    py::gil_scoped_release nogil;
    for (int w = 0; w < nworkers_; ++w) {
      if (byw[w].empty()) continue;
      {
        std::lock_guard<std::mutex> lk(ws_[w]->mu);
        for (auto& r : byw[w]) ws_[w]->responses.push_back(std::move(r));
      }
      ding(*ws_[w]);
    }
  }

  void set_labels(const std::vector<std::string>& labels,
                  const std::string& model) {
    labels_ = labels;
    model_ = model;
  }

  // Build the /predict JSON for a whole batch in C++ (the Python
  // responder's per-row json.dumps measured ~1.4 ms per 64-batch of
  // pure-GIL time) and enqueue the responses. topk/probs: [n][k].
  void respond_topk(const std::vector<uint64_t>& ids,
                    py::array_t<int32_t> topk,
                    py::array_t<float> probs) {
    auto ti = topk.unchecked<2>();
    auto pi = probs.unchecked<2>();
    const size_t n = ids.size();
    const py::ssize_t k = ti.shape(1);
    std::vector<Resp> byw[16];
    for (size_t i = 0; i < n; ++i) {
      std::string body;
      body.reserve(64 + 64 * k);
      body += "{\"predictions\":[";
      for (py::ssize_t j = 0; j < k; ++j) {
        int cls = ti(i, j);
        if (j) body += ',';
        body += "{\"label\":\"";
        if (cls >= 0 && (size_t)cls < labels_.size())
          json_escape(labels_[cls], body);
        body += "\",\"class_id\":";
        body += std::to_string(cls);
        char buf[32];
        snprintf(buf, sizeof(buf), ",\"prob\":%.6g}", (double)pi(i, j));
        body += buf;
      }
      body += "],\"model\":\"";
      json_escape(model_, body);
      body += "\"}";
      uint64_t id = ids[i];
      int w = (int)(id >> 60);
      if (w >= nworkers_) continue;
      byw[w].push_back(Resp{id, 200, std::move(body)});
    }
    py::gil_scoped_release nogil;
    for (int w = 0; w < nworkers_; ++w) {
      if (byw[w].empty()) continue;
      {
        std::lock_guard<std::mutex> lk(ws_[w]->mu);
        for (auto& r : byw[w]) ws_[w]->responses.push_back(std::move(r));
      }
      ding(*ws_[w]);
    }
  }

  py::dict stats() {
    py::dict d;
    d["accepted"] = accepted_.load();
    d["requests"] = requests_.load();
    d["responded"] = responded_.load();
    d["rejected_503"] = rejected_.load();
    d["dropped"] = dropped_.load();
    d["queue"] = [this] {
      std::lock_guard<std::mutex> lk(qmu_);
      return reqq_.size();
    }();
    return d;
  }

 private:
  struct Worker {
    int epfd = -1, evfd = -1;
    std::thread th;
    std::mutex mu;  // guards the two inboxes below
    std::vector<int> new_conns;
    std::vector<Resp> responses;
    std::unordered_map<int, std::unique_ptr<Conn>> conns;  // worker-local
    uint32_t gen_ctr = 1;
  };

  static void ding(Worker& w) {
    uint64_t one = 1;
    ssize_t r = ::write(w.evfd, &one, sizeof(one));
    (void)r;
  }

  uint64_t make_id(int widx, const Conn& c) const {
    return ((uint64_t)widx << 60) |
           ((uint64_t)(c.gen & 0xFFFFFFFFu) << 24) |
           (uint64_t)(c.fd & 0xFFFFFF);
  }

  void set_events(Worker& w, Conn& c) {
    epoll_event ev{};
    ev.events = (c.rd_closed ? 0u : (EPOLLIN | EPOLLRDHUP)) |
                (c.out_off < c.out.size() ? EPOLLOUT : 0u);
    ev.data.fd = c.fd;
    ::epoll_ctl(w.epfd, EPOLL_CTL_MOD, c.fd, &ev);
  }

  void close_conn(Worker& w, int fd) {
    auto it = w.conns.find(fd);
    if (it == w.conns.end()) return;
    ::epoll_ctl(w.epfd, EPOLL_CTL_DEL, fd, nullptr);
    ::close(fd);
    w.conns.erase(it);
  }

  bool flush(Worker& w, Conn& c) {  // false -> conn closed
    while (c.out_off < c.out.size()) {
      ssize_t n = ::send(c.fd, c.out.data() + c.out_off,
                         c.out.size() - c.out_off, MSG_NOSIGNAL);
      if (n > 0) {
        c.out_off += (size_t)n;
      } else if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
        break;
      } else {
        close_conn(w, c.fd);
        return false;
      }
    }
    if (c.out_off >= c.out.size()) {
      c.out.clear();
      c.out_off = 0;
      if (c.closing) {
        close_conn(w, c.fd);
        return false;
      }
    }
    return true;
  }

  void send_inline(Worker& w, Conn& c, int status, const std::string& body) {
    c.out += http_response(status, body, c.keep_alive);
    if (!c.keep_alive) c.closing = true;
  }

  // Parse as many complete requests as the lockstep policy allows.
  // Returns false if the conn was closed.
  bool parse_requests(int widx, Worker& w, Conn& c) {
    while (!c.awaiting) {
      size_t hdr_end = c.in.find("\r\n\r\n");
      if (hdr_end == std::string::npos) {
        if (c.in.size() > 64 * 1024) {  // runaway header block
          send_inline(w, c, 400, "{\"error\":\"bad request\"}");
          c.closing = true;
          break;
        }
        return flush(w, c);
      }
      size_t line_end = c.in.find("\r\n");
      std::string line = c.in.substr(0, line_end);
      size_t sp1 = line.find(' '), sp2 = line.rfind(' ');
      if (sp1 == std::string::npos || sp2 <= sp1) {
        close_conn(w, c.fd);
        return false;
      }
      std::string method = line.substr(0, sp1);
      std::string path = line.substr(sp1 + 1, sp2 - sp1 - 1);
      size_t qpos = path.find('?');
      if (qpos != std::string::npos) path.resize(qpos);

      size_t content_len = 0;
      bool keep_alive = line.find("HTTP/1.1") != std::string::npos;
      bool expect_100 = false;
      bool chunked = false;
      std::string boundary;  // multipart/form-data (the upload form)
      for (size_t p = line_end + 2; p < hdr_end;) {
        size_t e = c.in.find("\r\n", p);
        if (e == std::string::npos || e > hdr_end) e = hdr_end;
        size_t colon = c.in.find(':', p);
        if (colon != std::string::npos && colon < e) {
          const char* name = c.in.data() + p;
          size_t nlen = colon - p;
          size_t v = colon + 1;
          while (v < e && c.in[v] == ' ') ++v;
          if (nlen == 14 && iequal(name, "content-length", 14)) {
            content_len = (size_t)strtoull(c.in.c_str() + v, nullptr, 10);
          } else if (nlen == 10 && iequal(name, "connection", 10)) {
            keep_alive = !iequal(c.in.c_str() + v, "close", 5);
          } else if (nlen == 6 && iequal(name, "expect", 6)) {
            expect_100 = true;
          } else if (nlen == 17 &&
                     iequal(name, "transfer-encoding", 17)) {
            chunked = true;  // not supported: answered with 411 below
          } else if (nlen == 12 && iequal(name, "content-type", 12)) {
            std::string val = c.in.substr(v, e - v);
            size_t bp = val.find("boundary=");
            if (bp != std::string::npos &&
                val.compare(0, 19, "multipart/form-data") == 0) {
              boundary = val.substr(bp + 9);
              if (!boundary.empty() && boundary.front() == '"')
                boundary = boundary.substr(1, boundary.size() - 2);
            }
          }
        }
        p = e + 2;
      }
      c.keep_alive = keep_alive;

      if (content_len > kMaxBody) {
        send_inline(w, c, 413, "{\"error\":\"payload too large\"}");
        c.closing = true;
        break;
      }
      if (chunked) {
        send_inline(w, c, 411,
                    "{\"error\":\"length required (no chunked)\"}");
        c.closing = true;
        break;
      }
      size_t body_start = hdr_end + 4;
      if (c.in.size() < body_start + content_len) {
        if (expect_100 && !c.sent_100) {
          c.sent_100 = true;
          c.out += "HTTP/1.1 100 Continue\r\n\r\n";
        }
        return flush(w, c);  // need more body bytes
      }
      c.sent_100 = false;

      if (method == "POST" &&
          (path == "/predict" || path == "/api/predict")) {
        std::string body = c.in.substr(body_start, content_len);
        if (!boundary.empty()) {
          // extract the first file part's payload (what the Flask
          // surface does with request.files['file'])
          std::string delim = "--" + boundary;
          size_t p0 = body.find(delim);
          std::string part;
          while (p0 != std::string::npos) {
            size_t hdr_beg = p0 + delim.size();
            size_t hdr_end2 = body.find("\r\n\r\n", hdr_beg);
            if (hdr_end2 == std::string::npos) break;
            size_t nxt = body.find(delim, hdr_end2);
            if (nxt == std::string::npos) break;
            std::string phdr = body.substr(hdr_beg, hdr_end2 - hdr_beg);
            size_t data_beg = hdr_end2 + 4;
            size_t data_end = nxt >= 2 ? nxt - 2 : nxt;  // strip CRLF
            if (phdr.find("filename=") != std::string::npos ||
                part.empty())
              part = body.substr(data_beg, data_end - data_beg);
            if (phdr.find("filename=") != std::string::npos) break;
            p0 = nxt;
          }
          if (!part.empty()) body = std::move(part);
        }
        c.in.erase(0, body_start + content_len);
        requests_.fetch_add(1);
        bool enq = false;
        {
          std::lock_guard<std::mutex> lk(qmu_);
          if ((int)reqq_.size() < backlog_) {
            reqq_.push_back(Req{make_id(widx, c), std::move(body)});
            enq = true;
          }
        }
        if (enq) {
          qcv_.notify_all();
          c.awaiting = true;  // lockstep: respond before next parse
        } else {
          rejected_.fetch_add(1);
          send_inline(w, c, 503, "{\"error\":\"overloaded\"}");
        }
      } else {
        c.in.erase(0, body_start + content_len);
        if (method == "GET" && (path == "/" || path == "/index.html")) {
          // reference-parity upload form (the Flask app serves the
          // full template; this is the fast path's minimal equivalent)
          std::string html =
              "<!doctype html><title>predict</title><h3>image "
              "classification</h3><form method=post action=/predict "
              "enctype=multipart/form-data><input type=file "
              "name=file><input type=submit></form>";
          c.out += "HTTP/1.1 200 OK\r\nContent-Type: text/html\r\n"
                   "Content-Length: " +
                   std::to_string(html.size()) +
                   (c.keep_alive ? "\r\nConnection: keep-alive\r\n\r\n"
                                 : "\r\nConnection: close\r\n\r\n") +
              html;
          if (!c.keep_alive) c.closing = true;
        } else if (method == "GET" && path == "/healthz") {
          send_inline(w, c, 200, "{\"status\":\"ok\"}");
        } else if (method == "GET" && path == "/stats") {
          std::string s = "{\"requests\":" +
                          std::to_string(requests_.load()) +
                          ",\"responded\":" +
                          std::to_string(responded_.load()) + "}";
          send_inline(w, c, 200, s);
        } else if (method == "GET" && path == "/metrics") {
          // prometheus exposition (parity with the Flask /metrics)
          std::string m;
          m += "# TYPE twd_requests_total counter\n";
          m += "twd_requests_total " +
               std::to_string(requests_.load()) + "\n";
          m += "# TYPE twd_responses_total counter\n";
          m += "twd_responses_total " +
               std::to_string(responded_.load()) + "\n";
          m += "# TYPE twd_rejected_total counter\n";
          m += "twd_rejected_total " +
               std::to_string(rejected_.load()) + "\n";
          m += "# TYPE twd_connections_accepted_total counter\n";
          m += "twd_connections_accepted_total " +
               std::to_string(accepted_.load()) + "\n";
          c.out += "HTTP/1.1 200 OK\r\nContent-Type: text/plain; "
                   "version=0.0.4\r\nContent-Length: " +
                   std::to_string(m.size()) +
                   (c.keep_alive
                        ? "\r\nConnection: keep-alive\r\n\r\n"
                        : "\r\nConnection: close\r\n\r\n") +
               m;
          if (!c.keep_alive) c.closing = true;
        } else {
          send_inline(w, c, 404, "{\"error\":\"not found\"}");
        }
      }
    }
    return flush(w, c);
  }

  void worker_loop(int widx) {
    Worker& w = *ws_[widx];
    epoll_event evs[kMaxEvents];
    int64_t next_sweep = now_ms() + 5000;
    while (running_.load()) {
      int n = ::epoll_wait(w.epfd, evs, kMaxEvents, 200);
      if (n < 0) {
        if (errno == EINTR) continue;
        break;
      }
      const int64_t now = now_ms();
      if (now >= next_sweep) {
        // reap connections idle past the timeout UNLESS a request is
        // in flight (its response is still coming from Python)
        next_sweep = now + 5000;
        std::vector<int> dead;
        for (auto& kv : w.conns) {
          Conn& c = *kv.second;
          if (!c.awaiting && now - c.last_ms > idle_timeout_ms_)
            dead.push_back(kv.first);
        }
        for (int fd : dead) close_conn(w, fd);
      }
      for (int i = 0; i < n; ++i) {
        int fd = evs[i].data.fd;
        uint32_t e = evs[i].events;
        if (fd == w.evfd) {
          uint64_t junk;
          while (::read(w.evfd, &junk, sizeof(junk)) > 0) {
          }
          drain_inbox(widx, w);
        } else if (fd == listen_fd_) {
          accept_loop();
        } else {
          auto it = w.conns.find(fd);
          if (it == w.conns.end()) continue;
          Conn& c = *it->second;
          if (e & (EPOLLHUP | EPOLLERR)) {
            close_conn(w, fd);
            continue;
          }
          bool alive = true;
          c.last_ms = now;
          if (e & EPOLLIN) {
            char buf[64 * 1024];
            for (;;) {
              ssize_t r = ::recv(fd, buf, sizeof(buf), 0);
              if (r > 0) {
                c.in.append(buf, (size_t)r);
                if (r < (ssize_t)sizeof(buf)) break;
              } else if (r == 0) {
                // peer sent FIN: if a request is in flight keep the
                // conn for the response (half-close), else close
                c.rd_closed = true;
                if (!c.awaiting && c.out_off >= c.out.size()) {
                  close_conn(w, fd);
                  alive = false;
                }
                break;
              } else if (errno == EAGAIN || errno == EWOULDBLOCK) {
                break;
              } else {
                close_conn(w, fd);
                alive = false;
                break;
              }
            }
            if (alive) alive = parse_requests(widx, w, c);
          }
          if (alive && (e & EPOLLOUT)) alive = flush(w, c);
          if (alive) set_events(w, c);
        }
      }
    }
    // teardown: close everything this worker owns
    for (auto& kv : w.conns) ::close(kv.first);
    w.conns.clear();
    ::close(w.epfd);
    ::close(w.evfd);
  }

  void accept_loop() {
    for (;;) {
      int fd = ::accept4(listen_fd_, nullptr, nullptr,
                         SOCK_NONBLOCK | SOCK_CLOEXEC);
      if (fd < 0) break;
      int one = 1;
      ::setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      accepted_.fetch_add(1);
      int widx = (int)(accepted_.load() % (uint64_t)nworkers_);
      Worker& w = *ws_[widx];
      {
        std::lock_guard<std::mutex> lk(w.mu);
        w.new_conns.push_back(fd);
      }
      ding(w);
    }
  }

  void drain_inbox(int widx, Worker& w) {
    std::vector<int> nc;
    std::vector<Resp> rs;
    {
      std::lock_guard<std::mutex> lk(w.mu);
      nc.swap(w.new_conns);
      rs.swap(w.responses);
    }
    for (int fd : nc) {
      auto c = std::make_unique<Conn>();
      c->fd = fd;
      c->gen = w.gen_ctr++;
      c->last_ms = now_ms();
      epoll_event ev{};
      ev.events = EPOLLIN | EPOLLRDHUP;
      ev.data.fd = fd;
      if (::epoll_ctl(w.epfd, EPOLL_CTL_ADD, fd, &ev) == 0)
        w.conns.emplace(fd, std::move(c));
      else
        ::close(fd);
    }
    for (auto& r : rs) {
      int fd = (int)(r.id & 0xFFFFFF);
      uint32_t gen = (uint32_t)((r.id >> 24) & 0xFFFFFFFFu);
      auto it = w.conns.find(fd);
      if (it == w.conns.end() || it->second->gen != gen) {
        dropped_.fetch_add(1);  // client went away before the answer
        continue;
      }
      Conn& c = *it->second;
      c.out += http_response(r.status, r.body, c.keep_alive);
      if (!c.keep_alive || c.rd_closed) c.closing = true;
      c.awaiting = false;
      responded_.fetch_add(1);
      if (parse_requests(widx, w, c)) set_events(w, c);
    }
  }

  static int64_t now_ms() {
    return std::chrono::duration_cast<std::chrono::milliseconds>(
               std::chrono::steady_clock::now().time_since_epoch())
        .count();
  }

  int listen_fd_ = -1;
  int port_ = 0;
  int nworkers_;
  int backlog_;
  int64_t idle_timeout_ms_;
  std::vector<std::unique_ptr<Worker>> ws_;
  std::atomic<bool> running_{false};

  std::mutex qmu_;
  std::condition_variable qcv_;
  std::deque<Req> reqq_;

  std::atomic<uint64_t> accepted_{0}, requests_{0}, responded_{0},
      rejected_{0}, dropped_{0};
  std::vector<std::string> labels_;
  std::string model_;

  static void json_escape(const std::string& in, std::string& out) {
    for (char c : in) {
      if (c == '"' || c == '\\') {
        out += '\\';
        out += c;
      } else if ((unsigned char)c < 0x20) {
        char buf[8];
        snprintf(buf, sizeof(buf), "\\u%04x", c);
        out += buf;
      } else {
        out += c;
      }
    }
  }
};

}  // namespace

PYBIND11_MODULE(_httpd, m) {
  m.doc() = "epoll HTTP frontend: request handling without the GIL";
  py::class_<Frontend>(m, "Frontend")
      .def(py::init<const std::string&, int, int, int, int>(),
           py::arg("host"), py::arg("port"), py::arg("workers") = 3,
           py::arg("backlog") = 8192, py::arg("idle_timeout_s") = 120)
      .def("port", &Frontend::port)
      .def("wait_batch", &Frontend::wait_batch, py::arg("max_batch"),
           py::arg("max_wait_ms"))
      .def("respond_many", &Frontend::respond_many)
      .def("set_labels", &Frontend::set_labels)
      .def("respond_topk", &Frontend::respond_topk)
      .def("stats", &Frontend::stats)
      .def("stop", &Frontend::stop);
}
