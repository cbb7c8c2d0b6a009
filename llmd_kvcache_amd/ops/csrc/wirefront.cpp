// wirefront.cpp - native HTTP service front for the KV-cache indexer.
//
// Why C++: the fused kernel scores ~21M prompts/s but a Python HTTP
// stack tops out around 10k RPC/s per process - the wire, not the GPU,
// was the round-1 bottleneck.  This front keeps the
// per-request path entirely native (epoll + HTTP/1.1 keep-alive &
// pipelining + a handwritten JSON parser for the fixed request schema)
// and crosses into Python exactly ONCE per micro-batch, handing the
// scorer two flat tensors (tokens + offsets).  The Python side runs the
// same tested chain + fused-score path the bench measures.
//
// Reference analog: examples/kv_events/online/main.go:269-365 (the
// shipped Go HTTP binary).  Endpoints:
//   POST /score         {"model": m, "tokens": [...], "pods": [...]?}
//                    or {"model": m, "prompt": "...", "pods": [...]?}
//   GET  /health        {"status":"ok"}
// Response: {"scores":{"pod":score,...}}
//
// Concurrency model:
//  - N io threads, each with its own SO_REUSEPORT listener + epoll;
//    connections stay on their accepting thread;
//  - parsed requests enqueue into one MPSC batch queue; the batcher
//    drains whatever is queued (self-adjusting micro-batching, no
//    artificial window), groups by (model, pod filter, text/tokens),
//    calls the Python scorer per group, then hands each connection its
//    responses; per-connection slot ordering preserves HTTP/1.1
//    pipeline semantics;
//  - io threads never touch Python; the batcher acquires the GIL once
//    per group.

#include <arpa/inet.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/epoll.h>
#include <sys/eventfd.h>
#include <sys/socket.h>
#include <strings.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <list>
#include <map>
#include <unordered_map>
#include <memory>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include <torch/extension.h>

namespace kvidx {

// cpu_ops.cpp / hip_ops.hip (same extension)
std::vector<at::Tensor> hash_chain_batch(at::Tensor, at::Tensor, at::Tensor,
                                         int64_t);
at::Tensor cpu_fused_score(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                           at::Tensor, at::Tensor, at::Tensor, int64_t,
                           at::Tensor, at::Tensor, int64_t, at::Tensor,
                           at::Tensor, int64_t, int64_t);
#ifdef KVIDX_WITH_HIP
at::Tensor gpu_fused_score(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                           at::Tensor, at::Tensor, at::Tensor, int64_t,
                           at::Tensor, at::Tensor, int64_t, at::Tensor,
                           at::Tensor, int64_t, int64_t, int64_t, int64_t);
std::vector<at::Tensor> gpu_lookup(at::Tensor, at::Tensor, at::Tensor,
                                   at::Tensor, at::Tensor, at::Tensor,
                                   at::Tensor, int64_t, at::Tensor, int64_t,
                                   at::Tensor, int64_t, int64_t, int64_t,
                                   int64_t, int64_t);
at::Tensor gpu_score_from_masks(at::Tensor, at::Tensor, at::Tensor, int64_t);
#endif

namespace wire {

namespace py = pybind11;

// ---------------------------------------------------------------------
// Minimal JSON parsing for the fixed request schema.
// ---------------------------------------------------------------------

struct JsonCursor {
  const char* p;
  const char* end;
  bool ok = true;

  void skip_ws() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r'))
      ++p;
  }
  bool eat(char c) {
    skip_ws();
    if (p < end && *p == c) {
      ++p;
      return true;
    }
    ok = false;
    return false;
  }
  bool peek(char c) {
    skip_ws();
    return p < end && *p == c;
  }
};

// Parses a JSON string (with escapes) into out.  Cursor must be at '"'.
static bool parse_json_string(JsonCursor& c, std::string& out) {
  if (!c.eat('"')) return false;
  out.clear();
  while (c.p < c.end) {
    char ch = *c.p++;
    if (ch == '"') return true;
    if (ch != '\\') {
      out.push_back(ch);
      continue;
    }
    if (c.p >= c.end) break;
    char esc = *c.p++;
    switch (esc) {
      case '"': out.push_back('"'); break;
      case '\\': out.push_back('\\'); break;
      case '/': out.push_back('/'); break;
      case 'b': out.push_back('\b'); break;
      case 'f': out.push_back('\f'); break;
      case 'n': out.push_back('\n'); break;
      case 'r': out.push_back('\r'); break;
      case 't': out.push_back('\t'); break;
      case 'u': {
        if (c.end - c.p < 4) { c.ok = false; return false; }
        unsigned cp = 0;
        for (int i = 0; i < 4; ++i) {
          char h = *c.p++;
          cp <<= 4;
          if (h >= '0' && h <= '9') cp |= (unsigned)(h - '0');
          else if (h >= 'a' && h <= 'f') cp |= (unsigned)(h - 'a' + 10);
          else if (h >= 'A' && h <= 'F') cp |= (unsigned)(h - 'A' + 10);
          else { c.ok = false; return false; }
        }
        // surrogate pairs
        if (cp >= 0xD800 && cp <= 0xDBFF && c.end - c.p >= 6 &&
            c.p[0] == '\\' && c.p[1] == 'u') {
          unsigned lo = 0;
          const char* q = c.p + 2;
          bool good = true;
          for (int i = 0; i < 4; ++i) {
            char h = q[i];
            lo <<= 4;
            if (h >= '0' && h <= '9') lo |= (unsigned)(h - '0');
            else if (h >= 'a' && h <= 'f') lo |= (unsigned)(h - 'a' + 10);
            else if (h >= 'A' && h <= 'F') lo |= (unsigned)(h - 'A' + 10);
            else { good = false; break; }
          }
          if (good && lo >= 0xDC00 && lo <= 0xDFFF) {
            cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
            c.p += 6;
          }
        }
        // UTF-8 encode
        if (cp < 0x80) {
          out.push_back((char)cp);
        } else if (cp < 0x800) {
          out.push_back((char)(0xC0 | (cp >> 6)));
          out.push_back((char)(0x80 | (cp & 0x3F)));
        } else if (cp < 0x10000) {
          out.push_back((char)(0xE0 | (cp >> 12)));
          out.push_back((char)(0x80 | ((cp >> 6) & 0x3F)));
          out.push_back((char)(0x80 | (cp & 0x3F)));
        } else {
          out.push_back((char)(0xF0 | (cp >> 18)));
          out.push_back((char)(0x80 | ((cp >> 12) & 0x3F)));
          out.push_back((char)(0x80 | ((cp >> 6) & 0x3F)));
          out.push_back((char)(0x80 | (cp & 0x3F)));
        }
        break;
      }
      default:
        c.ok = false;
        return false;
    }
  }
  c.ok = false;
  return false;
}

static bool parse_int_array(JsonCursor& c, std::vector<int64_t>& out) {
  if (!c.eat('[')) return false;
  out.clear();
  c.skip_ws();
  if (c.peek(']')) {
    c.eat(']');
    return true;
  }
  while (c.p < c.end) {
    c.skip_ws();
    bool neg = false;
    if (c.p < c.end && *c.p == '-') {
      neg = true;
      ++c.p;
    }
    int64_t v = 0;
    bool any = false;
    while (c.p < c.end && *c.p >= '0' && *c.p <= '9') {
      v = v * 10 + (*c.p - '0');
      ++c.p;
      any = true;
    }
    if (!any) {
      c.ok = false;
      return false;
    }
    out.push_back(neg ? -v : v);
    c.skip_ws();
    if (c.p < c.end && *c.p == ',') {
      ++c.p;
      continue;
    }
    if (c.p < c.end && *c.p == ']') {
      ++c.p;
      return true;
    }
    break;
  }
  c.ok = false;
  return false;
}

static bool parse_string_array(JsonCursor& c, std::vector<std::string>& out) {
  if (!c.eat('[')) return false;
  out.clear();
  c.skip_ws();
  if (c.peek(']')) {
    c.eat(']');
    return true;
  }
  while (c.p < c.end) {
    std::string s;
    if (!parse_json_string(c, s)) return false;
    out.push_back(std::move(s));
    c.skip_ws();
    if (c.p < c.end && *c.p == ',') {
      ++c.p;
      continue;
    }
    if (c.p < c.end && *c.p == ']') {
      ++c.p;
      return true;
    }
    break;
  }
  c.ok = false;
  return false;
}

// Skips any JSON value (for unknown keys).
static bool skip_json_value(JsonCursor& c) {
  c.skip_ws();
  if (c.p >= c.end) {
    c.ok = false;
    return false;
  }
  char ch = *c.p;
  if (ch == '"') {
    std::string tmp;
    return parse_json_string(c, tmp);
  }
  if (ch == '{' || ch == '[') {
    char open = ch, close = (ch == '{') ? '}' : ']';
    int depth = 0;
    bool in_str = false;
    while (c.p < c.end) {
      char x = *c.p++;
      if (in_str) {
        if (x == '\\' && c.p < c.end) ++c.p;
        else if (x == '"') in_str = false;
        continue;
      }
      if (x == '"') in_str = true;
      else if (x == open) ++depth;
      else if (x == close && --depth == 0) return true;
    }
    c.ok = false;
    return false;
  }
  // number / literal
  while (c.p < c.end && *c.p != ',' && *c.p != '}' && *c.p != ']') ++c.p;
  return true;
}

// 128-bit prompt fingerprint: two independent FNV-1a streams over
// (model \x00 prompt).  Collision odds at the cache cap are ~2^-90;
// a collision would mis-route one scoring decision, never corrupt state.
struct PromptKey {
  uint64_t a, b;
  bool operator==(const PromptKey& o) const { return a == o.a && b == o.b; }
};
struct PromptKeyHash {
  size_t operator()(const PromptKey& k) const { return (size_t)(k.a ^ k.b); }
};
static inline void fp_mix_word(uint64_t w, uint64_t& h1, uint64_t& h2) {
  h1 = (h1 ^ w) * 0x9E3779B185EBCA87ull;
  h1 ^= h1 >> 31;
  h2 = (h2 ^ (w * 0xC2B2AE3D27D4EB4Full)) * 0x165667B19E3779F9ull;
  h2 ^= h2 >> 29;
}
static void fp_mix_bytes(const char* p, size_t n, uint64_t& h1,
                         uint64_t& h2) {
  // word-at-a-time (a byte loop costs ~3 ns/B, which at 45 KB prompts
  // was ~130 us/request - the text-mode wire bottleneck)
  size_t i = 0;
  for (; i + 8 <= n; i += 8) {
    uint64_t w;
    memcpy(&w, p + i, 8);
    fp_mix_word(w, h1, h2);
  }
  if (i < n) {
    uint64_t w = 0;
    memcpy(&w, p + i, n - i);
    fp_mix_word(w | ((uint64_t)(n - i) << 56), h1, h2);
  }
}
static PromptKey prompt_fingerprint(const std::string& model,
                                    const std::string& prompt) {
  uint64_t h1 = 14695981039346656037ull;
  uint64_t h2 = 0x9e3779b97f4a7c15ull;
  fp_mix_bytes(model.data(), model.size(), h1, h2);
  fp_mix_word(0xFFull ^ ((uint64_t)model.size() << 8), h1, h2);
  fp_mix_bytes(prompt.data(), prompt.size(), h1, h2);
  fp_mix_word((uint64_t)prompt.size(), h1, h2);
  return {h1, h2};
}

// Bounded LRU of prompt -> token ids (the C++ analog, one level up, of
// the reference's prefix store: a warm session prompt skips Python and
// the tokenizer entirely and rides the pre-tokenized batch path).
class PromptCache {
 public:
  explicit PromptCache(size_t max_entries) : cap_(max_entries) {}

  bool get(const PromptKey& k, std::vector<int64_t>& out) {
    std::lock_guard<std::mutex> lk(mu_);
    auto it = map_.find(k);
    if (it == map_.end()) return false;
    lru_.splice(lru_.begin(), lru_, it->second.first);
    const auto& v = it->second.second;
    out.assign(v.begin(), v.end());
    ++hits_;
    return true;
  }

  void put(const PromptKey& k, const int32_t* tok, size_t n) {
    std::lock_guard<std::mutex> lk(mu_);
    auto it = map_.find(k);
    if (it != map_.end()) {
      lru_.splice(lru_.begin(), lru_, it->second.first);
      return;
    }
    lru_.push_front(k);
    map_.emplace(k, std::make_pair(lru_.begin(),
                                   std::vector<int32_t>(tok, tok + n)));
    while (map_.size() > cap_) {
      map_.erase(lru_.back());
      lru_.pop_back();
    }
  }

  uint64_t hits() const {
    std::lock_guard<std::mutex> lk(mu_);
    return hits_;
  }

 private:
  size_t cap_;
  mutable std::mutex mu_;
  std::list<PromptKey> lru_;
  std::unordered_map<PromptKey,
                     std::pair<std::list<PromptKey>::iterator,
                               std::vector<int32_t>>,
                     PromptKeyHash>
      map_;
  uint64_t hits_ = 0;
};

struct ScoreRequest {
  std::string model;
  std::string prompt;           // text mode
  std::vector<int64_t> tokens;  // token mode
  std::vector<std::string> pods;
  bool has_tokens = false;
  bool has_prompt = false;
  bool want_cache = false;      // miss: store tokens after tokenization
  PromptKey cache_key{0, 0};
};

static bool parse_score_request(const char* body, size_t len,
                                ScoreRequest& out) {
  JsonCursor c{body, body + len};
  if (!c.eat('{')) return false;
  if (c.peek('}')) {
    c.eat('}');
    return false;  // empty request is invalid
  }
  while (c.ok) {
    std::string key;
    if (!parse_json_string(c, key)) return false;
    if (!c.eat(':')) return false;
    if (key == "model") {
      if (!parse_json_string(c, out.model)) return false;
    } else if (key == "prompt") {
      if (!parse_json_string(c, out.prompt)) return false;
      out.has_prompt = true;
    } else if (key == "tokens") {
      if (!parse_int_array(c, out.tokens)) return false;
      out.has_tokens = true;
    } else if (key == "pods") {
      if (!parse_string_array(c, out.pods)) return false;
    } else {
      if (!skip_json_value(c)) return false;
    }
    c.skip_ws();
    if (c.p < c.end && *c.p == ',') {
      ++c.p;
      continue;
    }
    if (c.p < c.end && *c.p == '}') return out.has_tokens || out.has_prompt;
    break;
  }
  return false;
}

// ---------------------------------------------------------------------
// Connections and io threads
// ---------------------------------------------------------------------

struct Conn {
  int fd = -1;
  int io_idx = 0;
  std::string inbuf;
  std::string outbuf;  // bytes ready to flush (io thread only)
  // pipelined response ordering: slot i must be sent before slot i+1
  std::mutex mu;
  uint64_t next_slot = 0;       // next slot id to assign (io thread)
  uint64_t flushed_slot = 0;    // next slot id to flush
  std::map<uint64_t, std::string> ready;  // completed out-of-order
  std::atomic<bool> closed{false};
  bool want_writable = false;
  bool sent_100 = false;  // io thread only
};

using ConnPtr = std::shared_ptr<Conn>;

struct PendingReq {
  ConnPtr conn;
  uint64_t slot;
  ScoreRequest req;
};

static void set_nonblock(int fd) {
  int fl = fcntl(fd, F_GETFL, 0);
  fcntl(fd, F_SETFL, fl | O_NONBLOCK);
}

// Pod identifiers are normally DNS-label-safe, but the event stream is
// external input - escape so a hostile name cannot break the JSON.
static void json_escape_into(std::string& out, const std::string& s) {
  for (unsigned char c : s) {
    switch (c) {
      case '"': out += "\\\""; break;
      case '\\': out += "\\\\"; break;
      case '\b': out += "\\b"; break;
      case '\f': out += "\\f"; break;
      case '\n': out += "\\n"; break;
      case '\r': out += "\\r"; break;
      case '\t': out += "\\t"; break;
      default:
        if (c < 0x20) {
          char buf[8];
          snprintf(buf, sizeof(buf), "\\u%04x", c);
          out += buf;
        } else {
          out += (char)c;
        }
    }
  }
}

static std::string http_response(int status, const char* status_text,
                                 const std::string& body) {
  std::string r;
  r.reserve(body.size() + 128);
  r += "HTTP/1.1 ";
  r += std::to_string(status);
  r += " ";
  r += status_text;
  r += "\r\ncontent-type: application/json\r\ncontent-length: ";
  r += std::to_string(body.size());
  r += "\r\n\r\n";
  r += body;
  return r;
}

class WireFront {
 public:
  WireFront(py::object score_tokens_cb, py::object score_text_cb,
            int64_t max_batch, int64_t n_batchers)
      : score_tokens_cb_(std::move(score_tokens_cb)),
        score_text_cb_(std::move(score_text_cb)),
        max_batch_(max_batch > 0 ? (size_t)max_batch : 4096),
        n_batchers_(n_batchers > 0 ? (int)n_batchers : 1) {}

  ~WireFront() {
    // The batcher threads acquire the GIL per batch; joining them while
    // holding it would deadlock.  stop() releases it explicitly, but a
    // destructor reached via garbage collection runs WITH the GIL held.
    if (PyGILState_Check()) {
      py::gil_scoped_release rel;
      stop_nogil();
    } else {
      stop_nogil();
    }
  }

  int start(const std::string& host, int port, int n_io) {
    TORCH_CHECK(!running_.load(), "wirefront already running");
    if (n_io < 1) n_io = 1;
    running_.store(true);
    bound_port_ = 0;
    for (int i = 0; i < n_io; ++i) {
      int lfd = socket(AF_INET, SOCK_STREAM, 0);
      TORCH_CHECK(lfd >= 0, "socket() failed");
      int one = 1;
      setsockopt(lfd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
      setsockopt(lfd, SOL_SOCKET, SO_REUSEPORT, &one, sizeof(one));
      sockaddr_in addr{};
      addr.sin_family = AF_INET;
      addr.sin_port = htons(bound_port_ ? bound_port_ : (uint16_t)port);
      addr.sin_addr.s_addr =
          host.empty() ? htonl(INADDR_LOOPBACK) : inet_addr(host.c_str());
      TORCH_CHECK(bind(lfd, (sockaddr*)&addr, sizeof(addr)) == 0,
                  "bind failed: ", strerror(errno));
      if (bound_port_ == 0) {
        socklen_t alen = sizeof(addr);
        getsockname(lfd, (sockaddr*)&addr, &alen);
        bound_port_ = ntohs(addr.sin_port);
      }
      TORCH_CHECK(listen(lfd, 1024) == 0, "listen failed");
      set_nonblock(lfd);
      listeners_.push_back(lfd);
      wakeups_.push_back(eventfd(0, EFD_NONBLOCK));
    }
    for (int i = 0; i < n_io; ++i)
      io_threads_.emplace_back([this, i] { io_loop(i); });
    // Multiple batchers overlap: while one is inside the GIL-free
    // scoring op (wire_score_flat-based callbacks release the GIL),
    // another drains/parses the next micro-batch.
    for (int b = 0; b < n_batchers_; ++b)
      batchers_.emplace_back([this] { batch_loop(); });
    return bound_port_;
  }

  void stop() {
    py::gil_scoped_release rel;
    stop_nogil();
  }

  int port() const { return bound_port_; }
  uint64_t requests() const { return n_requests_.load(); }
  uint64_t batches() const { return n_batches_.load(); }
  uint64_t prompt_cache_hits() const { return prompt_cache_.hits(); }

 private:
  void stop_nogil() {
    if (!running_.exchange(false)) return;
    queue_cv_.notify_all();
    for (int fd : listeners_) ::shutdown(fd, SHUT_RDWR);
    for (int efd : wakeups_) {
      uint64_t one = 1;
      (void)!write(efd, &one, sizeof(one));
    }
    for (auto& t : io_threads_)
      if (t.joinable()) t.join();
    for (auto& t : batchers_)
      if (t.joinable()) t.join();
    batchers_.clear();
    for (int fd : listeners_) close(fd);
    for (int efd : wakeups_) close(efd);
    listeners_.clear();
    wakeups_.clear();
    io_threads_.clear();
  }

  // ---- io thread ----------------------------------------------------
  void io_loop(int idx) {
    int ep = epoll_create1(0);
    epoll_event ev{};
    ev.events = EPOLLIN;
    ev.data.u64 = 1;  // listener tag
    epoll_ctl(ep, EPOLL_CTL_ADD, listeners_[idx], &ev);
    ev.events = EPOLLIN;
    ev.data.u64 = 2;  // wakeup tag
    epoll_ctl(ep, EPOLL_CTL_ADD, wakeups_[idx], &ev);

    std::vector<epoll_event> events(256);
    // fd -> conn for this thread
    std::map<int, ConnPtr> conns;

    while (running_.load()) {
      int n = epoll_wait(ep, events.data(), (int)events.size(), 200);
      for (int i = 0; i < n; ++i) {
        uint64_t tag = events[i].data.u64;
        if (tag == 1) {  // listener
          while (true) {
            int cfd = accept(listeners_[idx], nullptr, nullptr);
            if (cfd < 0) break;
            set_nonblock(cfd);
            int one = 1;
            setsockopt(cfd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
            auto conn = std::make_shared<Conn>();
            conn->fd = cfd;
            conn->io_idx = idx;
            conns[cfd] = conn;
            epoll_event cev{};
            cev.events = EPOLLIN;
            cev.data.fd = cfd;
            epoll_ctl(ep, EPOLL_CTL_ADD, cfd, &cev);
          }
          continue;
        }
        if (tag == 2) {  // wakeup: responses ready
          uint64_t drained;
          while (read(wakeups_[idx], &drained, sizeof(drained)) > 0) {
          }
          std::vector<ConnPtr> dirty;
          {
            std::lock_guard<std::mutex> lk(dirty_mu_[idx]);
            dirty.swap(dirty_conns_[idx]);
          }
          for (auto& c : dirty) {
            if (c->closed.load()) continue;
            collect_ready(*c);
            flush(ep, conns, c.get());
          }
          continue;
        }
        int fd = events[i].data.fd;
        auto it = conns.find(fd);
        if (it == conns.end()) continue;
        // hold a strong ref: flush() may close_conn(), which erases the
        // map entry (invalidating `it`) - the local ConnPtr keeps the
        // object alive and `closed` gates further use
        ConnPtr cp = it->second;
        if (events[i].events & (EPOLLHUP | EPOLLERR)) {
          close_conn(ep, conns, fd);
          continue;
        }
        if (events[i].events & EPOLLOUT) {
          flush(ep, conns, cp.get());
          if (cp->closed.load()) continue;
        }
        if (events[i].events & EPOLLIN) {
          if (!read_and_parse(cp)) {
            close_conn(ep, conns, fd);
            continue;
          }
          flush(ep, conns, cp.get());
        }
      }
      if (!running_.load()) break;
    }
    for (auto& [fd, c] : conns) {
      c->closed.store(true);
      close(fd);
    }
    close(ep);
  }

  void close_conn(int ep, std::map<int, ConnPtr>& conns, int fd) {
    auto it = conns.find(fd);
    if (it == conns.end()) return;
    it->second->closed.store(true);
    epoll_ctl(ep, EPOLL_CTL_DEL, fd, nullptr);
    close(fd);
    conns.erase(it);
  }

  // Reads available bytes; parses as many complete requests as present.
  // Returns false to close the connection.  All parsed /score requests
  // are enqueued under ONE queue lock + one notify so a pipelined burst
  // reaches the batcher as a block, not a trickle of singles.
  bool read_and_parse(const ConnPtr& conn) {
    std::vector<PendingReq> parsed;
    bool keep = read_and_parse_inner(conn, parsed);
    if (!parsed.empty()) {
      {
        std::lock_guard<std::mutex> lk(queue_mu_);
        for (auto& pr : parsed) queue_.push_back(std::move(pr));
      }
      queue_cv_.notify_one();
    }
    return keep;
  }

  bool read_and_parse_inner(const ConnPtr& conn,
                            std::vector<PendingReq>& parsed) {
    Conn* c = conn.get();
    char buf[64 * 1024];
    while (true) {
      ssize_t r = recv(c->fd, buf, sizeof(buf), 0);
      if (r > 0) {
        c->inbuf.append(buf, (size_t)r);
        if (c->inbuf.size() > (64u << 20)) return false;  // 64 MB guard
        continue;
      }
      if (r == 0) return false;  // peer closed
      if (errno == EAGAIN || errno == EWOULDBLOCK) break;
      return false;
    }
    // parse complete requests
    while (true) {
      size_t hdr_end = c->inbuf.find("\r\n\r\n");
      if (hdr_end == std::string::npos) {
        if (c->inbuf.size() > (1u << 20)) return false;  // header too big
        return true;
      }
      // request line
      size_t line_end = c->inbuf.find("\r\n");
      std::string line = c->inbuf.substr(0, line_end);
      size_t sp1 = line.find(' ');
      size_t sp2 = line.find(' ', sp1 + 1);
      if (sp1 == std::string::npos || sp2 == std::string::npos) return false;
      std::string method = line.substr(0, sp1);
      std::string path = line.substr(sp1 + 1, sp2 - sp1 - 1);
      // content-length (+ curl-style Expect: 100-continue)
      size_t clen = 0;
      bool expect_100 = false;
      {
        const char* h = c->inbuf.c_str() + line_end + 2;
        const char* hend = c->inbuf.c_str() + hdr_end + 2;
        while (h < hend) {
          const char* eol = strstr(h, "\r\n");
          if (!eol || eol > hend) break;
          if ((eol - h) > 15 && strncasecmp(h, "content-length:", 15) == 0) {
            clen = (size_t)strtoull(h + 15, nullptr, 10);
            // clamp BEFORE any arithmetic: a hostile value near
            // SIZE_MAX would wrap hdr_end+4+clen and hand the JSON
            // cursor an out-of-bounds end pointer
            if (clen > (64u << 20)) {
              return false;  // reject oversized/overflowing bodies
            }
          } else if ((eol - h) > 7 &&
                     strncasecmp(h, "expect:", 7) == 0 &&
                     strcasestr(std::string(h, eol - h).c_str(),
                                "100-continue")) {
            expect_100 = true;
          }
          h = eol + 2;
        }
      }
      size_t total = hdr_end + 4 + clen;
      if (c->inbuf.size() < total) {
        // interim 100 unblocks curl-style clients that wait before
        // sending the body; only safe to write directly when no earlier
        // response can still be pending (otherwise order would break -
        // the client's 1s fallback covers that rare mix)
        if (expect_100 && !c->sent_100) {
          bool idle;
          {
            std::lock_guard<std::mutex> lk(c->mu);
            idle = c->ready.empty() && c->flushed_slot == c->next_slot;
          }
          if (idle && c->outbuf.empty()) {
            c->outbuf += "HTTP/1.1 100 Continue\r\n\r\n";
            c->sent_100 = true;
          }
        }
        return true;  // body incomplete
      }
      c->sent_100 = false;

      if (method == "GET" && path == "/health") {
        enqueue_inline(conn, http_response(200, "OK", "{\"status\":\"ok\"}"));
      } else if (method == "POST" && path == "/score") {
        ScoreRequest req;
        if (parse_score_request(c->inbuf.data() + hdr_end + 4, clen, req)) {
          if (req.has_prompt && !req.has_tokens) {
            // warm-session fast path: a cached prompt becomes a
            // pre-tokenized request right here in the io thread
            PromptKey pk = prompt_fingerprint(req.model, req.prompt);
            if (prompt_cache_.get(pk, req.tokens)) {
              req.has_tokens = true;
              req.has_prompt = false;
              req.prompt.clear();
            } else {
              req.cache_key = pk;
              req.want_cache = true;
            }
          }
          uint64_t slot = c->next_slot++;
          parsed.push_back(PendingReq{conn, slot, std::move(req)});
        } else {
          enqueue_inline(
              conn, http_response(400, "Bad Request",
                                  "{\"error\":\"invalid score request\"}"));
        }
      } else {
        enqueue_inline(conn, http_response(404, "Not Found",
                                           "{\"error\":\"not found\"}"));
      }
      c->inbuf.erase(0, total);
      if (c->inbuf.empty()) return true;
    }
  }

  // io-thread-local immediate response still honors slot order.
  void enqueue_inline(const ConnPtr& conn, std::string resp) {
    Conn* c = conn.get();
    uint64_t slot = c->next_slot++;
    {
      std::lock_guard<std::mutex> lk(c->mu);
      c->ready.emplace(slot, std::move(resp));
    }
    collect_ready(*c);
  }

  // moves in-order completed responses into outbuf (io thread only)
  void collect_ready(Conn& c) {
    std::lock_guard<std::mutex> lk(c.mu);
    auto it = c.ready.find(c.flushed_slot);
    while (it != c.ready.end()) {
      c.outbuf += it->second;
      c.ready.erase(it);
      ++c.flushed_slot;
      it = c.ready.find(c.flushed_slot);
    }
  }

  void flush(int ep, std::map<int, ConnPtr>& conns, Conn* c) {
    collect_ready(*c);
    if (c->outbuf.size() > (64u << 20)) {
      // slow-reader guard: a client that pipelines requests without
      // ever reading responses would grow outbuf unboundedly
      close_conn(ep, conns, c->fd);
      return;
    }
    while (!c->outbuf.empty()) {
      ssize_t w = send(c->fd, c->outbuf.data(), c->outbuf.size(),
                       MSG_NOSIGNAL);
      if (w > 0) {
        c->outbuf.erase(0, (size_t)w);
        continue;
      }
      if (errno == EAGAIN || errno == EWOULDBLOCK) {
        if (!c->want_writable) {
          epoll_event ev{};
          ev.events = EPOLLIN | EPOLLOUT;
          ev.data.fd = c->fd;
          epoll_ctl(ep, EPOLL_CTL_MOD, c->fd, &ev);
          c->want_writable = true;
        }
        return;
      }
      close_conn(ep, conns, c->fd);
      return;
    }
    if (c->want_writable) {
      epoll_event ev{};
      ev.events = EPOLLIN;
      ev.data.fd = c->fd;
      epoll_ctl(ep, EPOLL_CTL_MOD, c->fd, &ev);
      c->want_writable = false;
    }
  }

  // ---- batcher ------------------------------------------------------
  void batch_loop() {
    while (running_.load()) {
      std::vector<PendingReq> batch;
      {
        std::unique_lock<std::mutex> lk(queue_mu_);
        queue_cv_.wait(lk, [this] {
          return !queue_.empty() || !running_.load();
        });
        if (!running_.load() && queue_.empty()) return;
        size_t take = std::min(queue_.size(), max_batch_);
        batch.assign(std::make_move_iterator(queue_.begin()),
                     std::make_move_iterator(queue_.begin() + take));
        queue_.erase(queue_.begin(), queue_.begin() + take);
        // No artificial collect window: batching is equilibrium-driven
        // (batch size = arrival rate x batch processing time), which a
        // measured A/B showed beats a fixed wait - the window added
        // idle time whenever the arrival rate itself was the bound.
        // Pipelined bursts still arrive as blocks because the io
        // threads enqueue a whole parse pass under one lock.
      }
      if (batch.empty()) continue;
      n_batches_.fetch_add(1);
      n_requests_.fetch_add(batch.size());

      // group by (model, pods, text/tokens)
      std::map<std::string, std::vector<size_t>> groups;
      for (size_t i = 0; i < batch.size(); ++i) {
        auto& r = batch[i].req;
        std::string key = r.has_tokens ? "T\x1f" : "P\x1f";
        key += r.model;
        for (auto& p : r.pods) {
          key += '\x1f';
          key += p;
        }
        groups[key].push_back(i);
      }
      for (auto& [key, idxs] : groups) run_group(batch, idxs);
      // wake owning io threads
      notify_io(batch);
    }
    // drain remaining requests with 503s so clients are not left hanging
    std::vector<PendingReq> rest;
    {
      std::lock_guard<std::mutex> lk(queue_mu_);
      rest.assign(std::make_move_iterator(queue_.begin()),
                  std::make_move_iterator(queue_.end()));
      queue_.clear();
    }
    for (auto& pr : rest)
      deliver(pr, http_response(503, "Service Unavailable",
                                "{\"error\":\"shutting down\"}"));
    notify_io(rest);
  }

  void run_group(std::vector<PendingReq>& batch,
                 const std::vector<size_t>& idxs) {
    bool text_mode = batch[idxs[0]].req.has_prompt;
    const std::string& model = batch[idxs[0]].req.model;
    const std::vector<std::string>& pods = batch[idxs[0]].req.pods;

    at::Tensor scores;
    std::vector<std::string> names;
    bool ok = false;
    std::string err;
    {
      py::gil_scoped_acquire gil;
      try {
        py::tuple result;
        py::tuple pods_t(pods.size());
        for (size_t i = 0; i < pods.size(); ++i) pods_t[i] = pods[i];
        if (text_mode) {
          py::list prompts;
          for (size_t i : idxs) prompts.append(batch[i].req.prompt);
          result = score_text_cb_(model, pods_t, prompts).cast<py::tuple>();
          if (result.size() >= 4) {
            // (scores, names, tokens_flat_i32, offsets_i64): feed the
            // prompt cache so the NEXT request for these prompts takes
            // the pre-tokenized path without touching Python
            auto tf = result[2].cast<at::Tensor>().contiguous();
            auto of = result[3].cast<at::Tensor>().contiguous();
            const int32_t* tp = tf.data_ptr<int32_t>();
            const int64_t* op = of.data_ptr<int64_t>();
            for (size_t k = 0; k < idxs.size(); ++k) {
              auto& rq = batch[idxs[k]].req;
              if (rq.want_cache)
                prompt_cache_.put(rq.cache_key, tp + op[k],
                                  (size_t)(op[k + 1] - op[k]));
            }
          }
        } else {
          int64_t total = 0;
          for (size_t i : idxs) total += (int64_t)batch[i].req.tokens.size();
          auto tokens = at::empty({total}, at::kLong);
          auto offsets = at::empty({(int64_t)idxs.size() + 1}, at::kLong);
          int64_t* tp = tokens.data_ptr<int64_t>();
          int64_t* op = offsets.data_ptr<int64_t>();
          op[0] = 0;
          int64_t pos = 0;
          for (size_t k = 0; k < idxs.size(); ++k) {
            auto& tv = batch[idxs[k]].req.tokens;
            std::memcpy(tp + pos, tv.data(), tv.size() * sizeof(int64_t));
            pos += (int64_t)tv.size();
            op[k + 1] = pos;
          }
          result =
              score_tokens_cb_(model, pods_t, tokens, offsets)
                  .cast<py::tuple>();
        }
        scores = result[0].cast<at::Tensor>()
                     .to(at::kFloat).contiguous();
        names = result[1].cast<std::vector<std::string>>();
        // validate INSIDE the try: a malformed callback return must
        // become a 500, never an uncaught throw in the batcher thread
        if (scores.dim() != 2 ||
            scores.size(0) != (int64_t)idxs.size())
          throw std::runtime_error("scorer returned wrong shape");
        ok = true;
      } catch (const std::exception& e) {
        err = e.what();
      }
    }
    if (!ok) {
      std::string body = "{\"error\":\"scorer failed\"}";
      for (size_t i : idxs)
        deliver(batch[i],
                http_response(500, "Internal Server Error", body));
      return;
    }
    const float* sp = scores.data_ptr<float>();
    int64_t P = scores.size(1);
    char num[64];
    for (size_t k = 0; k < idxs.size(); ++k) {
      std::string body = "{\"scores\":{";
      bool first = true;
      const float* row = sp + (int64_t)k * P;
      for (int64_t p = 0; p < P && p < (int64_t)names.size(); ++p) {
        if (row[p] == 0.f) continue;
        if (!first) body += ',';
        first = false;
        body += '"';
        json_escape_into(body, names[p]);
        body += "\":";
        int len = snprintf(num, sizeof(num), "%g", row[p]);
        body.append(num, len);
      }
      body += "}}";
      deliver(batch[idxs[k]], http_response(200, "OK", body));
    }
  }

  void deliver(PendingReq& pr, std::string resp) {
    Conn* c = pr.conn.get();
    {
      std::lock_guard<std::mutex> lk(c->mu);
      c->ready.emplace(pr.slot, std::move(resp));
    }
    std::lock_guard<std::mutex> lk(dirty_mu_[c->io_idx]);
    dirty_conns_[c->io_idx].push_back(pr.conn);
  }

  void notify_io(const std::vector<PendingReq>& batch) {
    bool seen[64] = {false};
    for (auto& pr : batch) {
      int idx = pr.conn->io_idx;
      if (idx >= 0 && idx < 64 && !seen[idx]) {
        seen[idx] = true;
        uint64_t one = 1;
        (void)!write(wakeups_[idx], &one, sizeof(one));
      }
    }
  }

  py::object score_tokens_cb_;
  py::object score_text_cb_;
  size_t max_batch_;
  std::atomic<bool> running_{false};
  int bound_port_ = 0;
  std::vector<int> listeners_;
  std::vector<int> wakeups_;
  std::vector<std::thread> io_threads_;
  std::vector<std::thread> batchers_;
  int n_batchers_ = 1;

  std::mutex queue_mu_;
  std::condition_variable queue_cv_;
  std::deque<PendingReq> queue_;

  std::mutex dirty_mu_[64];
  std::vector<ConnPtr> dirty_conns_[64];

  std::atomic<uint64_t> n_requests_{0};
  std::atomic<uint64_t> n_batches_{0};
  PromptCache prompt_cache_{2048};
};

// One-call scoring path for wire batches: parallel CPU chain -> fused
// probe/score (GPU or CPU table) -> CPU scores.  Registered with the
// GIL RELEASED so a second batcher thread can overlap its own batch's
// parse/launch with this one's kernel sync - and so io threads' Python
// never stalls behind scoring.
at::Tensor wire_score_flat(at::Tensor keys, at::Tensor meta, at::Tensor stamp,
                           at::Tensor pods, at::Tensor e_keys,
                           at::Tensor e_meta, at::Tensor e_vals,
                           int64_t pods_per_key, at::Tensor tokens_flat,
                           at::Tensor offsets, int64_t model_id,
                           at::Tensor filter_words, at::Tensor weights,
                           int64_t num_pods, int64_t epoch,
                           int64_t init_hash_bits, int64_t block_size,
                           int64_t n_tiers) {
  int64_t B = offsets.numel() - 1;
  auto parents = at::full({B}, init_hash_bits, at::kLong);
  auto chained = hash_chain_batch(tokens_flat, offsets, parents, block_size);
  at::Tensor hashes = chained[0];
  at::Tensor chunk_off = chained[1];  // int64 [B+1]
  bool is_cuda = keys.is_cuda();
  if (!is_cuda) {
    auto counts = (chunk_off.slice(0, 1, B + 1) -
                   chunk_off.slice(0, 0, B)).to(at::kInt);
    return cpu_fused_score(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                           pods_per_key, hashes, counts, model_id,
                           filter_words, weights, num_pods, epoch);
  }
#ifdef KVIDX_WITH_HIP
  auto dev = keys.device();
  auto h_dev = hashes.to(dev, /*non_blocking=*/false);
  auto off_dev = chunk_off.to(at::kInt).to(dev);
  int64_t max_k = 0;
  {
    const int64_t* co = chunk_off.data_ptr<int64_t>();
    for (int64_t b = 0; b < B; ++b)
      max_k = std::max(max_k, co[b + 1] - co[b]);
  }
  if (max_k == 0)
    return at::zeros({B, num_pods}, at::kFloat);
  int64_t W = (num_pods + 63) / 64;
  at::Tensor scores;
  if (max_k * n_tiers * W * 8 <= 64 * 1024) {
    scores = gpu_fused_score(keys, meta, stamp, pods, e_keys, e_meta,
                             e_vals, pods_per_key, h_dev, off_dev, model_id,
                             filter_words, weights, num_pods, epoch, max_k,
                             n_tiers);
  } else {
    auto fm = gpu_lookup(keys, meta, stamp, pods, e_keys, e_meta, e_vals,
                         pods_per_key, h_dev, model_id, filter_words,
                         num_pods, epoch, 0, 1, n_tiers);
    scores = gpu_score_from_masks(fm[1].contiguous(), off_dev, weights,
                                  num_pods);
  }
  return scores.to(at::kCPU, /*non_blocking=*/false).contiguous();
#else
  TORCH_CHECK(false, "built without HIP but table is on GPU");
#endif
}

void register_wirefront(py::module_& m) {
  m.def("wire_score_flat", &wire_score_flat,
        py::call_guard<py::gil_scoped_release>());
  py::class_<WireFront, std::shared_ptr<WireFront>>(m, "WireFront")
      .def(py::init<py::object, py::object, int64_t, int64_t>(),
           py::arg("score_tokens_cb"), py::arg("score_text_cb"),
           py::arg("max_batch") = 4096, py::arg("n_batchers") = 2)
      .def("start", &WireFront::start, py::arg("host") = std::string(),
           py::arg("port") = 0, py::arg("n_io") = 2)
      .def("stop", &WireFront::stop)
      .def("port", &WireFront::port)
      .def("requests", &WireFront::requests)
      .def("batches", &WireFront::batches)
      .def("prompt_cache_hits", &WireFront::prompt_cache_hits);
}

}  // namespace wire
}  // namespace kvidx
