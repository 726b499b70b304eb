// rayfed_amd C++ transport core — the cross-silo hot path without asyncio.
//
// The reference's data plane is Python gRPC inside Ray actor processes
// (/root/reference/fed/proxy/grpc/grpc_proxy.py); this engine's default
// Python transport is framed asyncio TCP (~0.1 ms/op).  This module is the
// native tier (SURVEY.md §2.3 "C++ data plane for the proxy hot path"):
// persistent sockets, framing, routing, the receive mailbox and ack
// round-trips all run in C++ threads with the GIL released — Python only
// (de)serializes payloads at the edges.
//
// Wire format (little-endian), the "xfer" framing:
//   u64 total_len ‖ u64 req_id ‖ u8 flags ‖ u8 job_len ‖ u8 up_len ‖
//   u8 down_len ‖ job ‖ up ‖ down ‖ body (opaque to C++)
//   ack: u32 len ‖ u64 req_id ‖ u16 code ‖ result (utf-8)
// flags bit 0 (DEFER_ACK): consume via the Python callback BEFORE acking
// (the shm lane's ack licenses segment recycling).
//
// TLS (OpenSSL, mutual auth optional) is supported on both roles.  The
// plaintext client pipelines requests on shared sockets (reader thread per
// connection); the TLS client instead uses a small pool of exclusive
// connections (one write+ack exchange at a time per connection) because an
// OpenSSL SSL* is not safe for a concurrent reader+writer pair.
// Thread-per-connection server: a federation has a handful of parties.
#include <openssl/err.h>
#include <openssl/ssl.h>
#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <sys/uio.h>
#include <unistd.h>

#include <atomic>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <map>
#include <mutex>
#include <optional>
#include <stdexcept>
#include <string>
#include <thread>
#include <unordered_map>
#include <unordered_set>
#include <vector>

namespace py = pybind11;

namespace {

constexpr uint8_t kFlagDeferAck = 1;
// STRIPE sub-frame: one shard of a large body, sent over its own
// connection in parallel with its siblings.  The preamble gains
// u32 idx ‖ u32 cnt ‖ u64 off ‖ u64 body_total after the seq ids; the
// receiver's connection threads read each stripe's payload DIRECTLY into
// the shared assembly buffer at `off` (no reassembly copy) and the last
// stripe posts the completed body to the mailbox.
constexpr uint8_t kFlagStripe = 2;
// PINNED body: assemble this (striped) frame into hipHostMalloc'd memory
// and deliver it to Python as a zero-copy view (wait_view) — the consumer
// H2Ds straight out of it, no bytes copy, no pinned staging bounce.
constexpr uint8_t kFlagPinned = 4;
// Reserved seq id of the init-time readiness barrier (constants.PING_SEQ_ID):
// acked without parking so repeated pings never leak mailbox slots.
constexpr const char* kPingSeqId = "ping";

// hipHostMalloc/hipHostFree, linked from amdhip64; declared here so this
// translation unit needs no HIP headers (plain g++ build).
extern "C" int hipHostMalloc(void** ptr, size_t size, unsigned int flags);
extern "C" int hipHostFree(void* ptr);

// Pooled pinned host buffers (power-of-two classes).  hipHostMalloc costs
// ~10 ms/100 MiB (page-locking), so chunk-streamed receives reuse buffers
// across chunks and transfers.  Falls back to plain malloc when no GPU
// runtime is live (CPU boxes) — the consumer then stages through its own
// pinned pool as before.
struct PinnedBuf {
  char* p = nullptr;
  size_t cap = 0;
  size_t len = 0;  // valid bytes of the current body
  bool pinned = false;
  ~PinnedBuf();
};

class PinnedPool {
 public:
  static PinnedPool& inst() {
    static PinnedPool pool;
    return pool;
  }

  // Static-destruction order vs the HIP runtime is undefined at process
  // exit — deliberately leak pooled buffers instead of calling
  // hipHostFree after the runtime may have unloaded.
  ~PinnedPool() {
    for (auto& [cls, bucket] : free_)
      for (auto& b : bucket) b->p = nullptr;
  }

  std::shared_ptr<PinnedBuf> acquire(size_t n) {
    size_t cls = 1ull << (64 - __builtin_clzll(std::max<size_t>(n, 1 << 16) - 1));
    {
      std::lock_guard<std::mutex> lk(mu_);
      auto& bucket = free_[cls];
      if (!bucket.empty()) {
        auto b = bucket.back();
        bucket.pop_back();
        b->len = n;
        return b;
      }
    }
    auto b = std::make_shared<PinnedBuf>();
    void* p = nullptr;
    if (hipHostMalloc(&p, cls, 0) == 0 && p) {
      b->p = (char*)p;
      b->pinned = true;
    } else {
      b->p = (char*)malloc(cls);
      if (!b->p) throw std::bad_alloc();
      b->pinned = false;
    }
    b->cap = cls;
    b->len = n;
    return b;
  }

  void release(std::shared_ptr<PinnedBuf> b) {
    std::lock_guard<std::mutex> lk(mu_);
    auto& bucket = free_[b->cap];
    size_t limit = b->cap >= (64u << 20) ? 16 : 8;  // <= ~4 GiB pinned held
    if (bucket.size() < limit) bucket.push_back(std::move(b));
    // else: shared_ptr drops it; ~PinnedBuf frees
  }

 private:
  std::mutex mu_;
  std::unordered_map<size_t, std::vector<std::shared_ptr<PinnedBuf>>> free_;
};

PinnedBuf::~PinnedBuf() {
  if (!p) return;
  if (pinned)
    hipHostFree(p);
  else
    free(p);
}

// ---------------------------------------------------------------- utilities
static void write_all(int fd, const char* data, size_t n) {
  while (n) {
    ssize_t w = ::send(fd, data, n, MSG_NOSIGNAL);
    if (w < 0) {
      if (errno == EINTR) continue;
      throw std::runtime_error(std::string("send: ") + strerror(errno));
    }
    data += w;
    n -= (size_t)w;
  }
}

static void writev_all(int fd, std::vector<iovec> iov) {
  size_t idx = 0;
  while (idx < iov.size()) {
    ssize_t w = ::writev(fd, iov.data() + idx, (int)(iov.size() - idx));
    if (w < 0) {
      if (errno == EINTR) continue;
      throw std::runtime_error(std::string("writev: ") + strerror(errno));
    }
    size_t ww = (size_t)w;
    while (idx < iov.size() && ww >= iov[idx].iov_len) {
      ww -= iov[idx].iov_len;
      ++idx;
    }
    if (idx < iov.size() && ww) {
      iov[idx].iov_base = (char*)iov[idx].iov_base + ww;
      iov[idx].iov_len -= ww;
    }
  }
}

static bool read_all(int fd, char* data, size_t n) {
  while (n) {
    ssize_t r = ::recv(fd, data, n, 0);
    if (r < 0) {
      if (errno == EINTR) continue;
      return false;
    }
    if (r == 0) return false;  // peer closed
    data += r;
    n -= (size_t)r;
  }
  return true;
}

static void set_sock_opts(int fd) {
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
  int buf = 8 << 20;
  setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &buf, sizeof(buf));
  setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &buf, sizeof(buf));
}

// Bounded busy-wait window before parking on a condition variable; tunable
// via RAYFED_SPIN_US (0 disables).  Applied on the tiny-message hot paths
// only (mailbox fetch, ack wait) — bulk transfers park immediately.
static int spin_us() {
  static int v = [] {
    const char* e = getenv("RAYFED_SPIN_US");
    return e ? atoi(e) : 120;
  }();
  return v;
}

static std::string ssl_err() {
  char buf[256];
  ERR_error_string_n(ERR_get_error(), buf, sizeof(buf));
  return std::string(buf);
}

// Byte stream over a plain fd or an SSL session.
struct Stream {
  int fd = -1;
  SSL* ssl = nullptr;

  bool read_all(char* data, size_t n) {
    if (!ssl) return ::read_all(fd, data, n);
    while (n) {
      int r = SSL_read(ssl, data, (int)std::min(n, (size_t)1 << 20));
      if (r <= 0) return false;
      data += r;
      n -= (size_t)r;
    }
    return true;
  }

  void write_all(const char* data, size_t n) {
    if (!ssl) return ::write_all(fd, data, n);
    while (n) {
      int w = SSL_write(ssl, data, (int)std::min(n, (size_t)1 << 20));
      if (w <= 0) throw std::runtime_error("SSL_write: " + ssl_err());
      data += w;
      n -= (size_t)w;
    }
  }

  void write_iov(std::vector<iovec> iov) {
    if (!ssl) return ::writev_all(fd, std::move(iov));
    for (auto& v : iov) write_all((const char*)v.iov_base, v.iov_len);
  }

  void close_free() {
    if (ssl) {
      SSL_free(ssl);
      ssl = nullptr;
    }
    if (fd >= 0) {
      ::close(fd);
      fd = -1;
    }
  }
};

static SSL_CTX* make_server_ctx(const std::string& cert, const std::string& key,
                                const std::string& ca) {
  SSL_CTX* ctx = SSL_CTX_new(TLS_server_method());
  if (!ctx) throw std::runtime_error("SSL_CTX_new failed");
  SSL_CTX_set_min_proto_version(ctx, TLS1_2_VERSION);
  if (SSL_CTX_use_certificate_chain_file(ctx, cert.c_str()) != 1 ||
      SSL_CTX_use_PrivateKey_file(ctx, key.c_str(), SSL_FILETYPE_PEM) != 1) {
    SSL_CTX_free(ctx);
    throw std::runtime_error("server cert/key load failed: " + ssl_err());
  }
  if (!ca.empty()) {
    if (SSL_CTX_load_verify_locations(ctx, ca.c_str(), nullptr) != 1) {
      SSL_CTX_free(ctx);
      throw std::runtime_error("server CA load failed: " + ssl_err());
    }
    SSL_CTX_set_verify(
        ctx, SSL_VERIFY_PEER | SSL_VERIFY_FAIL_IF_NO_PEER_CERT, nullptr);
  }
  return ctx;
}

static SSL_CTX* make_client_ctx(const std::string& ca, const std::string& cert,
                                const std::string& key) {
  SSL_CTX* ctx = SSL_CTX_new(TLS_client_method());
  if (!ctx) throw std::runtime_error("SSL_CTX_new failed");
  SSL_CTX_set_min_proto_version(ctx, TLS1_2_VERSION);
  if (!ca.empty() && SSL_CTX_load_verify_locations(ctx, ca.c_str(), nullptr) != 1) {
    SSL_CTX_free(ctx);
    throw std::runtime_error("client CA load failed: " + ssl_err());
  }
  SSL_CTX_set_verify(ctx, SSL_VERIFY_PEER, nullptr);
  if (!cert.empty() && !key.empty()) {
    if (SSL_CTX_use_certificate_chain_file(ctx, cert.c_str()) != 1 ||
        SSL_CTX_use_PrivateKey_file(ctx, key.c_str(), SSL_FILETYPE_PEM) != 1) {
      SSL_CTX_free(ctx);
      throw std::runtime_error("client cert/key load failed: " + ssl_err());
    }
  }
  return ctx;
}

// Zero-copy Python view over a completed pinned body; returns the buffer
// to the pool when garbage-collected (after the consumer's H2D completes —
// the exporter stays alive while memoryviews over it exist).
class BodyView {
 public:
  explicit BodyView(std::shared_ptr<PinnedBuf> b) : buf_(std::move(b)) {}
  ~BodyView() {
    if (buf_) {
      try {
        PinnedPool::inst().release(std::move(buf_));
      } catch (...) {
      }
    }
  }
  char* data() const { return buf_->p; }
  size_t size() const { return buf_->len; }
  bool pinned() const { return buf_->pinned; }

 private:
  std::shared_ptr<PinnedBuf> buf_;
};

// ------------------------------------------------------------------- server
class XferServer {
 public:
  XferServer(int port, std::string job_name, std::string tls_cert = "",
             std::string tls_key = "", std::string tls_ca = "")
      : job_(std::move(job_name)), port_(port) {
    if (!tls_cert.empty()) {
      ssl_ctx_ = make_server_ctx(tls_cert, tls_key, tls_ca);
    }
  }

  ~XferServer() {
    stop();
    if (ssl_ctx_) SSL_CTX_free(ssl_ctx_);
  }

  // consume_cb(up, down, body_bytes, token) is called (with the GIL) for
  // DEFER_ACK frames only.  It must NOT block: it hands the decode to a
  // Python worker thread and returns immediately; the worker acks by calling
  // complete(token, code, result) when the consume (H2D + CRC) is done.
  // This keeps a multi-GiB consume from stalling the connection thread —
  // other frames on the same connection keep flowing while it runs.
  void start(py::object consume_cb) {
    consume_cb_ = std::move(consume_cb);
    listen_fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
    if (listen_fd_ < 0) throw std::runtime_error("socket() failed");
    int zero = 0, one = 1;
    // Parity with grpc.so_reuseport=0: exclusive bind, but allow
    // REUSEADDR so TIME_WAIT ports rebind across test runs.
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEPORT, &zero, sizeof(zero));
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_addr.s_addr = htonl(INADDR_ANY);
    addr.sin_port = htons((uint16_t)port_);
    if (bind(listen_fd_, (sockaddr*)&addr, sizeof(addr)) != 0) {
      ::close(listen_fd_);
      listen_fd_ = -1;
      throw std::runtime_error("bind failed: port in use");
    }
    if (listen(listen_fd_, 64) != 0) {
      ::close(listen_fd_);
      listen_fd_ = -1;
      throw std::runtime_error("listen failed");
    }
    running_ = true;
    accept_thread_ = std::thread([this] { accept_loop(); });
  }

  void stop() {
    bool expected = true;
    if (!running_.compare_exchange_strong(expected, false)) return;
    ::shutdown(listen_fd_, SHUT_RDWR);
    ::close(listen_fd_);
    {
      std::lock_guard<std::mutex> lk(conn_mu_);
      for (auto& c : conns_) ::shutdown(c->st.fd, SHUT_RDWR);
    }
    if (accept_thread_.joinable()) accept_thread_.join();
    {
      std::lock_guard<std::mutex> lk(conn_mu_);
      for (auto& t : conn_threads_)
        if (t.joinable()) t.join();
      conn_threads_.clear();
    }
    {  // outstanding deferred acks become no-ops
      std::lock_guard<std::mutex> lk(tok_mu_);
      tokens_.clear();
    }
    // Unblock any waiting get_data / wait_view.
    std::lock_guard<std::mutex> lk(mail_mu_);
    stopped_ = true;
    pinned_mail_.clear();
    mail_cv_.notify_all();
  }

  // Ack a deferred-consume frame from a Python worker thread (GIL released
  // by the binding).  No-op if the connection died or the server stopped.
  void complete(uint64_t token, int code, const std::string& result) {
    std::shared_ptr<SrvConn> conn;
    uint64_t req_id = 0;
    {
      std::lock_guard<std::mutex> lk(tok_mu_);
      auto it = tokens_.find(token);
      if (it == tokens_.end()) return;
      conn = it->second.first;
      req_id = it->second.second;
      tokens_.erase(it);
    }
    if (!conn->alive.load()) return;
    try {
      write_ack(*conn, req_id, (uint16_t)code, result);
    } catch (...) {
      conn->alive = false;
      ::shutdown(conn->st.fd, SHUT_RDWR);
    }
  }

  // Blocking fetch (GIL released by the binding); returns body bytes.
  // Spins briefly before the cv wait: on the tiny-task hot path the reply
  // lands within ~100 us, and catching it in the spin saves a futex
  // wake-from-idle (~50 us when the waiter has gone cold).
  py::bytes get_data(const std::string& up, const std::string& down,
                     double timeout_s) {
    std::string key = up + '\x00' + down;
    std::string body;
    {
      py::gil_scoped_release release;
      auto deadline = std::chrono::steady_clock::now() +
                      std::chrono::duration<double>(timeout_s);
      const auto spin_until = std::chrono::steady_clock::now() +
                              std::chrono::microseconds(spin_us());
      bool got = false;
      while (!got && std::chrono::steady_clock::now() < spin_until) {
        {
          std::unique_lock<std::mutex> lk(mail_mu_, std::try_to_lock);
          if (lk.owns_lock()) {
            auto it = mail_.find(key);
            if (it != mail_.end()) {
              body = std::move(it->second);
              mail_.erase(it);
              got = true;
              break;
            }
            if (stopped_) throw std::runtime_error("server stopped");
          }
        }
        for (int i = 0; i < 32; ++i) __builtin_ia32_pause();
      }
      if (!got) {
        std::unique_lock<std::mutex> lk(mail_mu_);
        while (true) {
          auto it = mail_.find(key);
          if (it != mail_.end()) {
            body = std::move(it->second);
            mail_.erase(it);
            break;
          }
          if (stopped_) throw std::runtime_error("server stopped");
          if (mail_cv_.wait_until(lk, deadline) == std::cv_status::timeout)
            throw std::runtime_error("get_data timeout");
        }
      }
    }
    return py::bytes(body);
  }

  // Insert a marker/body locally (used after a deferred consume so a
  // blocked get_data wakes and finds the Python-side decoded object).
  void post(const std::string& up, const std::string& down,
            const std::string& body) {
    std::lock_guard<std::mutex> lk(mail_mu_);
    mail_[up + '\x00' + down] = body;
    mail_cv_.notify_all();
  }

  // Pinned-body mail: blocking / non-blocking fetch of a completed
  // kFlagPinned body as a zero-copy BodyView.
  py::object wait_view(const std::string& up, const std::string& down,
                       double timeout_s) {
    std::string key = up + '\x00' + down;
    std::shared_ptr<PinnedBuf> b;
    {
      py::gil_scoped_release release;
      std::unique_lock<std::mutex> lk(mail_mu_);
      auto deadline = std::chrono::steady_clock::now() +
                      std::chrono::duration<double>(timeout_s);
      while (true) {
        auto it = pinned_mail_.find(key);
        if (it != pinned_mail_.end()) {
          b = std::move(it->second);
          pinned_mail_.erase(it);
          break;
        }
        if (stopped_) throw std::runtime_error("server stopped");
        if (mail_cv_.wait_until(lk, deadline) == std::cv_status::timeout)
          throw std::runtime_error("wait_view timeout");
      }
    }
    return py::cast(new BodyView(std::move(b)),
                    py::return_value_policy::take_ownership);
  }

  py::object try_view(const std::string& up, const std::string& down) {
    std::string key = up + '\x00' + down;
    std::shared_ptr<PinnedBuf> b;
    {
      std::lock_guard<std::mutex> lk(mail_mu_);
      auto it = pinned_mail_.find(key);
      if (it == pinned_mail_.end()) return py::none();
      b = std::move(it->second);
      pinned_mail_.erase(it);
    }
    return py::cast(new BodyView(std::move(b)),
                    py::return_value_policy::take_ownership);
  }

  // Non-blocking probe for the fast path.
  std::optional<py::bytes> try_take(const std::string& up,
                                    const std::string& down) {
    std::string key = up + '\x00' + down;
    std::string body;
    {
      py::gil_scoped_release release;
      std::lock_guard<std::mutex> lk(mail_mu_);
      auto it = mail_.find(key);
      if (it == mail_.end()) return std::nullopt;
      body = std::move(it->second);
      mail_.erase(it);
    }
    return py::bytes(body);
  }

  uint64_t received_op_count() const { return recv_count_.load(); }

 private:
  // A server-side connection: the conn thread reads frames; acks may be
  // written by the conn thread (inline path) or a Python worker completing
  // a deferred consume — write_mu keeps them from interleaving.
  struct SrvConn {
    Stream st;
    std::mutex write_mu;
    std::atomic<bool> alive{true};
  };

  // In-flight striped body, filled by several connection threads at once.
  // `got` counts bytes landed; the stripe that completes the byte count
  // moves the body to the mailbox (or, for kFlagPinned frames, to the
  // pinned-view mail).  (A retried frame after a partial failure recreates
  // the slot only if its geometry changed; a stale mix is caught by the
  // payload checksum downstream.)
  struct Assembly {
    std::string data;                 // plain bodies
    std::shared_ptr<PinnedBuf> pbuf;  // kFlagPinned bodies
    std::atomic<uint64_t> got{0};
    uint64_t total;
    uint32_t cnt;
    // Per-stripe arrival flags: a retried frame re-sends stripes that
    // already landed; counting them again would overshoot `got` and could
    // deliver with a range missing (or race a writer with the delivery
    // move).  Duplicates are drained instead.
    std::unique_ptr<std::atomic<uint8_t>[]> seen;
    Assembly(uint64_t t, uint32_t c, bool pinned)
        : total(t), cnt(c), seen(new std::atomic<uint8_t>[c]) {
      for (uint32_t i = 0; i < c; ++i) seen[i].store(0);
      if (pinned)
        pbuf = PinnedPool::inst().acquire(t);
      else
        data.resize(t);
    }
    char* base() { return pbuf ? pbuf->p : data.data(); }
  };

  static bool drain(Stream& st, uint64_t n) {
    char scratch[65536];
    while (n) {
      size_t k = (size_t)std::min<uint64_t>(n, sizeof(scratch));
      if (!st.read_all(scratch, k)) return false;
      n -= k;
    }
    return true;
  }

  static void write_ack(SrvConn& c, uint64_t req_id, uint16_t code,
                        const std::string& result) {
    char ack[14];
    uint32_t ack_len = 10 + (uint32_t)result.size();
    memcpy(ack, &ack_len, 4);
    memcpy(ack + 4, &req_id, 8);
    memcpy(ack + 12, &code, 2);
    std::lock_guard<std::mutex> lk(c.write_mu);
    // Re-check under the lock: teardown closes/frees the stream while
    // holding write_mu, so a dead conn is never written (no SSL* UAF).
    if (!c.alive.load()) throw std::runtime_error("connection closed");
    c.st.write_all(ack, 14);
    if (!result.empty()) c.st.write_all(result.data(), result.size());
  }

  void accept_loop() {
    while (running_) {
      int fd = ::accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) {
        if (running_ && errno == EINTR) continue;
        break;
      }
      set_sock_opts(fd);
      auto conn = std::make_shared<SrvConn>();
      conn->st.fd = fd;
      std::lock_guard<std::mutex> lk(conn_mu_);
      conns_.push_back(conn);
      conn_threads_.emplace_back([this, conn] { conn_loop(conn); });
    }
  }

  void conn_loop(std::shared_ptr<SrvConn> conn) {
    Stream& st = conn->st;
    if (ssl_ctx_) {
      st.ssl = SSL_new(ssl_ctx_);
      SSL_set_fd(st.ssl, st.fd);
      if (SSL_accept(st.ssl) != 1) {
        conn->alive = false;
        st.close_free();
        return;
      }
    }
    std::vector<char> buf;
    while (running_) {
      char head[16];
      if (!st.read_all(head, 16)) break;
      uint64_t total, req_id;
      memcpy(&total, head, 8);
      memcpy(&req_id, head + 8, 8);
      if (total < 12 || total > (64ull << 30)) break;  // sane bounds
      char meta[4];
      if (!st.read_all(meta, 4)) break;
      const uint8_t flags = (uint8_t)meta[0];
      const uint8_t job_len = (uint8_t)meta[1];
      const uint8_t up_len = (uint8_t)meta[2];
      const uint8_t down_len = (uint8_t)meta[3];
      const size_t names_len = (size_t)job_len + up_len + down_len;
      const size_t fixed =
          8 + 4 + names_len + ((flags & kFlagStripe) ? 24u : 0u);
      if (total < fixed) break;
      char names[768];
      if (!st.read_all(names, names_len)) break;
      std::string job(names, job_len);
      std::string up(names + job_len, up_len);
      std::string down(names + job_len + up_len, down_len);
      const uint64_t payload_len = total - fixed;

      uint16_t code = 200;
      std::string result = "OK";
      if (flags & kFlagStripe) {
        char smeta[24];
        if (!st.read_all(smeta, 24)) break;
        uint32_t idx, cnt;
        uint64_t s_off, body_total;
        memcpy(&idx, smeta, 4);
        memcpy(&cnt, smeta + 4, 4);
        memcpy(&s_off, smeta + 8, 8);
        memcpy(&body_total, smeta + 16, 8);
        if (cnt == 0 || idx >= cnt || body_total > (64ull << 30) ||
            s_off + payload_len > body_total)
          break;
        if (job != job_) {
          if (!drain(st, payload_len)) break;
          code = 417;
          result = "JobName mis-match: expected " + job_ + ", got " + job;
        } else {
          std::string key = up + '\x00' + down;
          const bool pinned = (flags & kFlagPinned) != 0;
          std::shared_ptr<Assembly> asmb;
          bool tombstoned = false;
          {
            std::lock_guard<std::mutex> lk(asm_mu_);
            auto& slot = asm_[key];
            if (!slot) {
              // A retried stripe can trail its frame's delivery; starting
              // a fresh assembly for it would leak a half-filled buffer.
              if (delivered_set_.count(key)) {
                asm_.erase(key);
                tombstoned = true;
              } else {
                slot = std::make_shared<Assembly>(body_total, cnt, pinned);
              }
            } else if (slot->total != body_total || slot->cnt != cnt) {
              slot = std::make_shared<Assembly>(body_total, cnt, pinned);
            }
            if (!tombstoned) asmb = slot;
          }
          if (tombstoned || asmb->seen[idx].exchange(1)) {
            // Duplicate stripe (whole-frame retry after one stripe
            // failed) — the bytes already landed; swallow this copy.
            if (!drain(st, payload_len)) break;
          } else
          // Payload streams DIRECTLY into the assembly buffer at its
          // offset — stripes of one frame write disjoint ranges from
          // their own connection threads, no reassembly copy.  For pinned
          // bodies the buffer is hipHostMalloc'd, so the socket reads land
          // in DMA-able memory.
          if (!st.read_all(asmb->base() + s_off, payload_len)) {
            asmb->seen[idx].store(0);  // not landed: the retry must write
            break;
          } else if (asmb->got.fetch_add(payload_len) + payload_len ==
                     body_total) {
            {
              std::lock_guard<std::mutex> lk(asm_mu_);
              asm_.erase(key);
              delivered_set_.insert(key);
              delivered_fifo_.push_back(key);
              if (delivered_fifo_.size() > 256) {
                delivered_set_.erase(delivered_fifo_.front());
                delivered_fifo_.pop_front();
              }
            }
            recv_count_.fetch_add(1);
            std::lock_guard<std::mutex> lk(mail_mu_);
            if (asmb->pbuf)
              pinned_mail_[key] = std::move(asmb->pbuf);
            else
              mail_[key] = std::move(asmb->data);
            mail_cv_.notify_all();
          }
        }
        try {
          write_ack(*conn, req_id, code, result);
        } catch (...) {
          break;
        }
        continue;
      }
      buf.resize(payload_len);
      if (!st.read_all(buf.data(), payload_len)) break;
      size_t off = 0;

      if (job != job_) {
        code = 417;
        result = "JobName mis-match: expected " + job_ + ", got " + job;
      } else if (up == kPingSeqId && down == kPingSeqId) {
        // Readiness ping: ack without parking (nothing ever consumes it)
        // and without counting it as a received data op.
      } else if (flags & kFlagDeferAck) {
        // shm/IPC-lane frame: Python consumes (H2D + CRC) before the ack —
        // hand it off and keep reading; the worker acks via complete().
        uint64_t token = next_token_.fetch_add(1);
        {
          std::lock_guard<std::mutex> lk(tok_mu_);
          tokens_[token] = {conn, req_id};
        }
        bool handed_off = false;
        {
          py::gil_scoped_acquire gil;
          try {
            py::bytes body(buf.data() + off, buf.size() - off);
            consume_cb_(up, down, body, token);
            handed_off = true;
          } catch (const std::exception& e) {
            code = 500;
            result = std::string("consume dispatch failed: ") + e.what();
          }
        }
        if (handed_off) continue;  // ack comes later from complete()
        std::lock_guard<std::mutex> lk(tok_mu_);
        tokens_.erase(token);
      } else {
        recv_count_.fetch_add(1);
        std::lock_guard<std::mutex> lk(mail_mu_);
        mail_[up + '\x00' + down] =
            std::string(buf.data() + off, buf.size() - off);
        mail_cv_.notify_all();
      }

      try {
        write_ack(*conn, req_id, code, result);
      } catch (...) {
        break;
      }
    }
    {
      std::lock_guard<std::mutex> lk(conn->write_mu);
      conn->alive = false;
      st.close_free();
    }
  }

  std::string job_;
  int port_;
  int listen_fd_ = -1;
  std::atomic<bool> running_{false};
  bool stopped_ = false;
  std::thread accept_thread_;
  std::mutex conn_mu_;
  std::vector<std::shared_ptr<SrvConn>> conns_;
  std::vector<std::thread> conn_threads_;
  std::mutex mail_mu_;
  std::condition_variable mail_cv_;
  std::map<std::string, std::string> mail_;
  std::mutex asm_mu_;
  std::unordered_map<std::string, std::shared_ptr<Assembly>> asm_;
  // Recently delivered stripe keys (guarded by asm_mu_): late duplicate
  // stripes of an already-delivered frame are drained, not reassembled.
  std::unordered_set<std::string> delivered_set_;
  std::deque<std::string> delivered_fifo_;
  std::unordered_map<std::string, std::shared_ptr<PinnedBuf>> pinned_mail_;
  std::atomic<uint64_t> recv_count_{0};
  std::mutex tok_mu_;
  std::unordered_map<uint64_t, std::pair<std::shared_ptr<SrvConn>, uint64_t>>
      tokens_;
  std::atomic<uint64_t> next_token_{1};
  py::object consume_cb_;
  SSL_CTX* ssl_ctx_ = nullptr;
};

// ------------------------------------------------------------------- client
class XferClient {
 public:
  XferClient(std::string job_name, std::string tls_ca = "",
             std::string tls_cert = "", std::string tls_key = "",
             std::string server_name = "")
      : job_(std::move(job_name)), server_name_(std::move(server_name)) {
    if (!tls_ca.empty() || !tls_cert.empty()) {
      ssl_ctx_ = make_client_ctx(tls_ca, tls_cert, tls_key);
    }
  }
  ~XferClient() {
    close_all();
    if (ssl_ctx_) SSL_CTX_free(ssl_ctx_);
  }

  // Slice a scatter-gather view list down to the byte range [lo, hi).
  static std::vector<std::pair<const char*, size_t>> slice_views(
      const std::vector<std::pair<const char*, size_t>>& views, uint64_t lo,
      uint64_t hi) {
    std::vector<std::pair<const char*, size_t>> out;
    uint64_t pos = 0;
    for (const auto& v : views) {
      uint64_t vlo = pos, vhi = pos + v.second;
      pos = vhi;
      if (vhi <= lo || vlo >= hi) continue;
      uint64_t s = std::max(vlo, lo), e = std::min(vhi, hi);
      out.emplace_back(v.first + (s - vlo), (size_t)(e - s));
    }
    return out;
  }

  // Striped send: split the body across `stripes` parallel connections —
  // a single TCP stream tops out ~2 GB/s on this path; N streams with the
  // receiver assembling in place scale it.  Blocks until every stripe is
  // acked; any stripe failure fails the send (caller retries whole-frame).
  int send_striped(const std::string& host, int port, const std::string& up,
                   const std::string& down,
                   const std::vector<std::pair<const char*, size_t>>& views,
                   uint64_t body_len, int stripes, double timeout_s,
                   std::string* result_out, bool pinned = false) {
    std::string names;
    names.push_back((char)(kFlagStripe | (pinned ? kFlagPinned : 0)));
    names.push_back((char)job_.size());
    names.push_back((char)up.size());
    names.push_back((char)down.size());
    names += job_;
    names += up;
    names += down;
    const uint64_t chunk = (body_len + stripes - 1) / stripes;
    std::vector<int> codes((size_t)stripes, 0);
    std::vector<std::string> results((size_t)stripes);
    std::vector<std::string> errs((size_t)stripes);
    std::vector<std::thread> threads;
    for (int i = 0; i < stripes; ++i) {
      threads.emplace_back([&, i] {
        try {
          const uint64_t lo = (uint64_t)i * chunk;
          const uint64_t hi = std::min(body_len, lo + chunk);
          auto sub = slice_views(views, lo, hi);
          char smeta[24];
          uint32_t idx = (uint32_t)i, cnt = (uint32_t)stripes;
          memcpy(smeta, &idx, 4);
          memcpy(smeta + 4, &cnt, 4);
          memcpy(smeta + 8, &lo, 8);
          memcpy(smeta + 16, &body_len, 8);
          uint64_t total = 8 + names.size() + 24 + (hi - lo);
          if (ssl_ctx_) {
            codes[i] = stripe_tls(host, port, names, smeta, sub, total,
                                  timeout_s, &results[i]);
            return;
          }
          std::shared_ptr<Conn> conn_sp = get_conn(host, port, i + 1);
          Conn& conn = *conn_sp;
          std::shared_ptr<Pending> pending = std::make_shared<Pending>();
          uint64_t req_id;
          {
            std::lock_guard<std::mutex> lk(conn.write_mu);
            req_id = conn.next_id++;
            {
              std::lock_guard<std::mutex> lk2(conn.pend_mu);
              conn.pending[req_id] = pending;
            }
            char head[16];
            memcpy(head, &total, 8);
            memcpy(head + 8, &req_id, 8);
            std::vector<iovec> iov;
            iov.push_back({head, 16});
            iov.push_back({(void*)names.data(), names.size()});
            iov.push_back({smeta, 24});
            for (auto& v : sub) iov.push_back({(void*)v.first, v.second});
            try {
              writev_all(conn.fd, std::move(iov));
            } catch (...) {
              conn.alive = false;
              throw;
            }
          }
          std::unique_lock<std::mutex> lk(pending->mu);
          auto deadline = std::chrono::steady_clock::now() +
                          std::chrono::duration<double>(timeout_s);
          while (!pending->done) {
            if (pending->cv.wait_until(lk, deadline) ==
                std::cv_status::timeout)
              throw std::runtime_error("stripe ack timeout");
          }
          if (pending->broken)
            throw std::runtime_error("connection broken awaiting stripe ack");
          codes[i] = pending->code;
          results[i] = pending->result;
        } catch (const std::exception& e) {
          errs[i] = e.what();
        }
      });
    }
    for (auto& t : threads) t.join();
    for (int i = 0; i < stripes; ++i)
      if (!errs[i].empty())
        throw std::runtime_error("stripe " + std::to_string(i) +
                                 " failed: " + errs[i]);
    for (int i = 0; i < stripes; ++i) {
      if (codes[i] != 200) {
        if (result_out) *result_out = results[i];
        return codes[i];
      }
    }
    if (result_out) *result_out = "OK";
    return 200;
  }

  int stripe_tls(const std::string& host, int port, const std::string& names,
                 const char* smeta,
                 const std::vector<std::pair<const char*, size_t>>& sub,
                 uint64_t total, double timeout_s, std::string* result_out) {
    std::string key = host + ":" + std::to_string(port);
    std::unique_ptr<Stream> st = acquire_tls_conn(key, host, port);
    set_recv_timeout(st->fd, std::max(1.0, timeout_s));
    uint64_t req_id = tls_req_id_.fetch_add(1);
    try {
      char head[16];
      memcpy(head, &total, 8);
      memcpy(head + 8, &req_id, 8);
      std::vector<iovec> iov;
      iov.push_back({head, 16});
      iov.push_back({(void*)names.data(), names.size()});
      iov.push_back({(void*)smeta, 24});
      for (auto& v : sub) iov.push_back({(void*)v.first, v.second});
      st->write_iov(std::move(iov));
      char ahead[14];
      if (!st->read_all(ahead, 14))
        throw std::runtime_error("connection broken awaiting stripe ack");
      uint32_t len;
      uint64_t rid;
      uint16_t code;
      memcpy(&len, ahead, 4);
      memcpy(&rid, ahead + 4, 8);
      memcpy(&code, ahead + 12, 2);
      std::string result;
      if (len > 10) {
        result.resize(len - 10);
        if (!st->read_all(result.data(), result.size()))
          throw std::runtime_error("connection broken reading stripe ack");
      }
      if (rid != req_id)
        throw std::runtime_error("ack id mismatch on TLS stripe");
      release_tls_conn(key, std::move(st));
      if (result_out) *result_out = result;
      return code;
    } catch (...) {
      st->close_free();
      throw;
    }
  }

  // Blocking send with ack round trip; GIL released around I/O.
  // parts: list of buffer-likes written scatter-gather (no join copy).
  // stripes > 1 splits the body across that many parallel connections.
  int send(const std::string& host, int port, const std::string& up,
           const std::string& down, std::vector<py::buffer> parts,
           bool defer_ack, double timeout_s, std::string* result_out,
           int stripes = 1, bool pinned = false) {
    // Collect buffer pointers under the GIL.
    std::vector<std::pair<const char*, size_t>> views;
    views.reserve(parts.size());
    size_t body_len = 0;
    std::vector<py::buffer_info> infos;
    infos.reserve(parts.size());
    for (auto& b : parts) {
      infos.emplace_back(b.request());
      auto& info = infos.back();
      views.emplace_back((const char*)info.ptr,
                         (size_t)(info.size * info.itemsize));
      body_len += views.back().second;
    }

    if (stripes > 1 && !defer_ack && body_len > 0) {
      py::gil_scoped_release release;
      return send_striped(host, port, up, down, views, body_len, stripes,
                          timeout_s, result_out, pinned);
    }

    uint8_t flags = defer_ack ? kFlagDeferAck : 0;
    std::string preamble;
    preamble.push_back((char)flags);
    preamble.push_back((char)job_.size());
    preamble.push_back((char)up.size());
    preamble.push_back((char)down.size());
    preamble += job_;
    preamble += up;
    preamble += down;

    uint64_t total = 8 /*req id*/ + preamble.size() + body_len;

    int code;
    std::string result;
    if (ssl_ctx_) {
      py::gil_scoped_release release;
      return send_tls(host, port, preamble, views, total, timeout_s,
                      result_out);
    }
    {
      py::gil_scoped_release release;
      std::shared_ptr<Conn> conn_sp = get_conn(host, port);
      Conn& conn = *conn_sp;  // shared ownership: safe vs concurrent replace
      uint64_t req_id;
      std::shared_ptr<Pending> pending = std::make_shared<Pending>();
      {
        std::lock_guard<std::mutex> lk(conn.write_mu);
        req_id = conn.next_id++;
        {
          std::lock_guard<std::mutex> lk2(conn.pend_mu);
          conn.pending[req_id] = pending;
        }
        char head[16];
        memcpy(head, &total, 8);
        memcpy(head + 8, &req_id, 8);
        std::vector<iovec> iov;
        iov.push_back({head, 16});
        iov.push_back({(void*)preamble.data(), preamble.size()});
        for (auto& v : views) iov.push_back({(void*)v.first, v.second});
        try {
          writev_all(conn.fd, std::move(iov));
        } catch (...) {
          conn.alive = false;
          throw;
        }
      }
      // Spin briefly for the ack (loopback RTT ~10-50 us) before parking.
      const auto spin_until = std::chrono::steady_clock::now() +
                              std::chrono::microseconds(spin_us());
      while (!pending->done.load(std::memory_order_acquire) &&
             std::chrono::steady_clock::now() < spin_until) {
        for (int i = 0; i < 32; ++i) __builtin_ia32_pause();
      }
      std::unique_lock<std::mutex> lk(pending->mu);
      auto deadline = std::chrono::steady_clock::now() +
                      std::chrono::duration<double>(timeout_s);
      while (!pending->done) {
        if (pending->cv.wait_until(lk, deadline) == std::cv_status::timeout) {
          throw std::runtime_error("ack timeout");
        }
      }
      if (pending->broken)
        throw std::runtime_error("connection broken awaiting ack");
      code = pending->code;
      result = pending->result;
    }
    if (result_out) *result_out = result;
    return code;
  }

  // Fire-and-wait-later send (plaintext lane): write the frame in the
  // CALLER's thread (it is already hot — typically the producer completing
  // the payload) and return a handle; wait_ack() collects the ack from
  // whichever thread tracks it.  Eliminates the send-pool wake-from-idle
  // (~60-150 us measured) on the tiny-task critical path.
  uint64_t send_async(const std::string& host, int port, const std::string& up,
                      const std::string& down, std::vector<py::buffer> parts,
                      bool defer_ack) {
    if (ssl_ctx_)
      throw std::runtime_error("send_async unsupported on the TLS lane");
    std::vector<std::pair<const char*, size_t>> views;
    views.reserve(parts.size());
    size_t body_len = 0;
    std::vector<py::buffer_info> infos;
    infos.reserve(parts.size());
    for (auto& b : parts) {
      infos.emplace_back(b.request());
      auto& info = infos.back();
      views.emplace_back((const char*)info.ptr,
                         (size_t)(info.size * info.itemsize));
      body_len += views.back().second;
    }
    uint8_t flags = defer_ack ? kFlagDeferAck : 0;
    std::string preamble;
    preamble.push_back((char)flags);
    preamble.push_back((char)job_.size());
    preamble.push_back((char)up.size());
    preamble.push_back((char)down.size());
    preamble += job_;
    preamble += up;
    preamble += down;
    uint64_t total = 8 + preamble.size() + body_len;

    std::shared_ptr<Pending> pending = std::make_shared<Pending>();
    {
      py::gil_scoped_release release;
      std::shared_ptr<Conn> conn_sp = get_conn(host, port);
      Conn& conn = *conn_sp;
      uint64_t req_id;
      std::lock_guard<std::mutex> lk(conn.write_mu);
      req_id = conn.next_id++;
      {
        std::lock_guard<std::mutex> lk2(conn.pend_mu);
        conn.pending[req_id] = pending;
      }
      char head[16];
      memcpy(head, &total, 8);
      memcpy(head + 8, &req_id, 8);
      std::vector<iovec> iov;
      iov.push_back({head, 16});
      iov.push_back({(void*)preamble.data(), preamble.size()});
      for (auto& v : views) iov.push_back({(void*)v.first, v.second});
      try {
        writev_all(conn.fd, std::move(iov));
      } catch (...) {
        conn.alive = false;
        throw;
      }
    }
    uint64_t h = async_id_.fetch_add(1);
    std::lock_guard<std::mutex> lk(async_mu_);
    async_[h] = pending;
    return h;
  }

  std::pair<int, std::string> wait_ack(uint64_t h, double timeout_s) {
    std::shared_ptr<Pending> pending;
    {
      std::lock_guard<std::mutex> lk(async_mu_);
      auto it = async_.find(h);
      if (it == async_.end())
        throw std::runtime_error("unknown ack handle");
      pending = it->second;
      async_.erase(it);
    }
    const auto spin_until = std::chrono::steady_clock::now() +
                            std::chrono::microseconds(spin_us());
    while (!pending->done.load(std::memory_order_acquire) &&
           std::chrono::steady_clock::now() < spin_until) {
      for (int i = 0; i < 32; ++i) __builtin_ia32_pause();
    }
    std::unique_lock<std::mutex> lk(pending->mu);
    auto deadline = std::chrono::steady_clock::now() +
                    std::chrono::duration<double>(timeout_s);
    while (!pending->done) {
      if (pending->cv.wait_until(lk, deadline) == std::cv_status::timeout)
        throw std::runtime_error("ack timeout");
    }
    if (pending->broken)
      throw std::runtime_error("connection broken awaiting ack");
    return {pending->code, pending->result};
  }

  static void set_recv_timeout(int fd, double timeout_s) {
    timeval tv;
    tv.tv_sec = (time_t)timeout_s;
    tv.tv_usec = (suseconds_t)((timeout_s - (double)tv.tv_sec) * 1e6);
    setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
  }

  int send_tls(const std::string& host, int port, const std::string& preamble,
               const std::vector<std::pair<const char*, size_t>>& views,
               uint64_t total, double timeout_s, std::string* result_out) {
    std::string key = host + ":" + std::to_string(port);
    std::unique_ptr<Stream> st = acquire_tls_conn(key, host, port);
    // Bound the ack wait: a half-open peer must surface as a broken
    // connection (retryable) instead of hanging the exchange forever.
    set_recv_timeout(st->fd, std::max(1.0, timeout_s));
    uint64_t req_id = tls_req_id_.fetch_add(1);
    try {
      char head[16];
      memcpy(head, &total, 8);
      memcpy(head + 8, &req_id, 8);
      std::vector<iovec> iov;
      iov.push_back({head, 16});
      iov.push_back({(void*)preamble.data(), preamble.size()});
      for (auto& v : views) iov.push_back({(void*)v.first, v.second});
      st->write_iov(std::move(iov));
      char ahead[14];
      if (!st->read_all(ahead, 14))
        throw std::runtime_error("connection broken awaiting ack");
      uint32_t len;
      uint64_t rid;
      uint16_t code;
      memcpy(&len, ahead, 4);
      memcpy(&rid, ahead + 4, 8);
      memcpy(&code, ahead + 12, 2);
      std::string result;
      if (len > 10) {
        result.resize(len - 10);
        if (!st->read_all(result.data(), result.size()))
          throw std::runtime_error("connection broken reading ack body");
      }
      if (rid != req_id) throw std::runtime_error("ack id mismatch on TLS lane");
      release_tls_conn(key, std::move(st));
      if (result_out) *result_out = result;
      return code;
    } catch (...) {
      st->close_free();
      throw;
    }
  }

  std::unique_ptr<Stream> acquire_tls_conn(const std::string& key,
                                           const std::string& host, int port) {
    {
      std::lock_guard<std::mutex> lk(tls_mu_);
      auto& pool = tls_pool_[key];
      if (!pool.empty()) {
        auto st = std::move(pool.back());
        pool.pop_back();
        return st;
      }
    }
    int fd = ::socket(AF_INET, SOCK_STREAM, 0);
    if (fd < 0) throw std::runtime_error("socket() failed");
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons((uint16_t)port);
    if (inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1) {
      hostent* he = gethostbyname(host.c_str());
      if (!he) {
        ::close(fd);
        throw std::runtime_error("resolve failed: " + host);
      }
      memcpy(&addr.sin_addr, he->h_addr, sizeof(addr.sin_addr));
    }
    if (connect(fd, (sockaddr*)&addr, sizeof(addr)) != 0) {
      ::close(fd);
      throw std::runtime_error(std::string("connect failed: ") +
                               strerror(errno));
    }
    set_sock_opts(fd);
    auto st = std::make_unique<Stream>();
    st->fd = fd;
    st->ssl = SSL_new(ssl_ctx_);
    SSL_set_fd(st->ssl, fd);
    // Verify against the destination host unless a name is pinned explicitly
    // (grpc.ssl_target_name_override semantics) — never a fixed default.
    const std::string& sni = server_name_.empty() ? host : server_name_;
    SSL_set_tlsext_host_name(st->ssl, sni.c_str());
    SSL_set1_host(st->ssl, sni.c_str());
    if (SSL_connect(st->ssl) != 1) {
      std::string e = ssl_err();
      st->close_free();
      throw std::runtime_error("TLS handshake failed: " + e);
    }
    return st;
  }

  void release_tls_conn(const std::string& key, std::unique_ptr<Stream> st) {
    std::lock_guard<std::mutex> lk(tls_mu_);
    auto& pool = tls_pool_[key];
    // Chunk-streamed sends fan out to chunk-workers x stripes concurrent
    // TLS exchanges (8 x 8 at the defaults); retaining fewer connections
    // than that forces fresh handshakes every round.
    if (pool.size() < 64) {
      pool.push_back(std::move(st));
    } else {
      st->close_free();
    }
  }

  void close_all() {
    std::unordered_map<std::string, std::shared_ptr<Conn>> conns;
    {
      std::lock_guard<std::mutex> lk(conns_mu_);
      conns.swap(conns_);
    }
    for (auto& [key, conn] : conns) {
      ::shutdown(conn->fd, SHUT_RDWR);
      if (conn->reader.joinable()) conn->reader.join();
      ::close(conn->fd);
    }
    std::lock_guard<std::mutex> lk2(tls_mu_);
    for (auto& [key, pool] : tls_pool_)
      for (auto& st : pool) st->close_free();
    tls_pool_.clear();
  }

 private:
  struct Pending {
    std::mutex mu;
    std::condition_variable cv;
    std::atomic<bool> done{false};  // atomic: spun on outside the lock
    bool broken = false;
    uint16_t code = 0;
    std::string result;
  };

  struct Conn {
    int fd = -1;
    uint64_t next_id = 1;
    std::mutex write_mu;
    std::mutex pend_mu;
    std::unordered_map<uint64_t, std::shared_ptr<Pending>> pending;
    std::thread reader;
    std::atomic<bool> alive{true};
  };

  std::shared_ptr<Conn> get_conn(const std::string& host, int port,
                                 int slot = 0) {
    std::string key = host + ":" + std::to_string(port);
    if (slot) key += "#" + std::to_string(slot);
    // Serialize (re)connection per client: the lock is held through
    // connect() so concurrent first-sends share one socket, and dead conns
    // are REPLACED, never destroyed, while a sender still references them
    // (shared_ptr ownership).
    std::lock_guard<std::mutex> lk(conns_mu_);
    auto it = conns_.find(key);
    if (it != conns_.end() && it->second->alive) return it->second;
    int fd = ::socket(AF_INET, SOCK_STREAM, 0);
    if (fd < 0) throw std::runtime_error("socket() failed");
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons((uint16_t)port);
    if (inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1) {
      hostent* he = gethostbyname(host.c_str());
      if (!he) {
        ::close(fd);
        throw std::runtime_error("resolve failed: " + host);
      }
      memcpy(&addr.sin_addr, he->h_addr, sizeof(addr.sin_addr));
    }
    if (connect(fd, (sockaddr*)&addr, sizeof(addr)) != 0) {
      ::close(fd);
      throw std::runtime_error(std::string("connect failed: ") +
                               strerror(errno));
    }
    set_sock_opts(fd);
    auto conn = std::make_shared<Conn>();
    conn->fd = fd;
    std::shared_ptr<Conn> sp = conn;
    conn->reader = std::thread([this, sp] { reader_loop(sp.get()); });
    auto& entry = conns_[key];
    if (entry) {  // replace a dead conn; readers/senders keep their refs
      ::shutdown(entry->fd, SHUT_RDWR);
      if (entry->reader.joinable()) entry->reader.detach();
      dead_.push_back(entry);
    }
    entry = conn;
    return conn;
  }

  void reader_loop(Conn* conn) {
    while (true) {
      char head[14];
      if (!read_all(conn->fd, head, 14)) break;
      uint32_t len;
      uint64_t req_id;
      uint16_t code;
      memcpy(&len, head, 4);
      memcpy(&req_id, head + 4, 8);
      memcpy(&code, head + 12, 2);
      std::string result;
      if (len > 10) {
        result.resize(len - 10);
        if (!read_all(conn->fd, result.data(), result.size())) break;
      }
      std::shared_ptr<Pending> p;
      {
        std::lock_guard<std::mutex> lk(conn->pend_mu);
        auto it = conn->pending.find(req_id);
        if (it != conn->pending.end()) {
          p = it->second;
          conn->pending.erase(it);
        }
      }
      if (p) {
        std::lock_guard<std::mutex> lk(p->mu);
        p->done = true;
        p->code = code;
        p->result = std::move(result);
        p->cv.notify_all();
      }
    }
    conn->alive = false;
    std::lock_guard<std::mutex> lk(conn->pend_mu);
    for (auto& [id, p] : conn->pending) {
      std::lock_guard<std::mutex> lk2(p->mu);
      p->done = true;
      p->broken = true;
      p->cv.notify_all();
    }
    conn->pending.clear();
  }

  std::string job_;
  std::string server_name_;
  SSL_CTX* ssl_ctx_ = nullptr;
  std::mutex conns_mu_;
  std::unordered_map<std::string, std::shared_ptr<Conn>> conns_;
  std::vector<std::shared_ptr<Conn>> dead_;  // kept until close_all
  std::mutex tls_mu_;
  std::unordered_map<std::string, std::vector<std::unique_ptr<Stream>>> tls_pool_;
  std::atomic<uint64_t> tls_req_id_{1};
  std::mutex async_mu_;
  std::unordered_map<uint64_t, std::shared_ptr<Pending>> async_;
  std::atomic<uint64_t> async_id_{1};
};

}  // namespace

PYBIND11_MODULE(_xfer, m) {
  m.doc() = "rayfed_amd C++ transport core (plaintext cross-silo hot path)";
  py::class_<BodyView>(m, "BodyView", py::buffer_protocol())
      .def_buffer([](BodyView& v) {
        return py::buffer_info(v.data(), sizeof(uint8_t),
                               py::format_descriptor<uint8_t>::format(), 1,
                               {v.size()}, {sizeof(uint8_t)});
      })
      .def("__len__", &BodyView::size)
      .def_property_readonly("pinned", &BodyView::pinned);
  py::class_<XferServer>(m, "XferServer")
      .def(py::init<int, std::string, std::string, std::string, std::string>(),
           py::arg("port"), py::arg("job_name"), py::arg("tls_cert") = "",
           py::arg("tls_key") = "", py::arg("tls_ca") = "")
      .def("start", &XferServer::start, py::arg("consume_cb"))
      .def("stop", &XferServer::stop,
           py::call_guard<py::gil_scoped_release>())
      .def("get_data", &XferServer::get_data, py::arg("up"), py::arg("down"),
           py::arg("timeout_s") = 600.0)
      .def("complete", &XferServer::complete, py::arg("token"),
           py::arg("code"), py::arg("result") = std::string(),
           py::call_guard<py::gil_scoped_release>())
      .def("post", &XferServer::post, py::call_guard<py::gil_scoped_release>())
      .def("wait_view", &XferServer::wait_view, py::arg("up"),
           py::arg("down"), py::arg("timeout_s") = 60.0)
      .def("try_view", &XferServer::try_view)
      .def("try_take", &XferServer::try_take)
      .def_property_readonly("received_op_count",
                             &XferServer::received_op_count);
  py::class_<XferClient>(m, "XferClient")
      .def(py::init<std::string, std::string, std::string, std::string,
                    std::string>(),
           py::arg("job_name"), py::arg("tls_ca") = "",
           py::arg("tls_cert") = "", py::arg("tls_key") = "",
           py::arg("server_name") = "")
      .def(
          "send",
          [](XferClient& c, const std::string& host, int port,
             const std::string& up, const std::string& down,
             std::vector<py::buffer> parts, bool defer_ack,
             double timeout_s, int stripes, bool pinned) {
            std::string result;
            int code =
                c.send(host, port, up, down, std::move(parts), defer_ack,
                       timeout_s, &result, stripes, pinned);
            return py::make_tuple(code, result);
          },
          py::arg("host"), py::arg("port"), py::arg("up"), py::arg("down"),
          py::arg("parts"), py::arg("defer_ack") = false,
          py::arg("timeout_s") = 60.0, py::arg("stripes") = 1,
          py::arg("pinned") = false)
      .def("send_async", &XferClient::send_async, py::arg("host"),
           py::arg("port"), py::arg("up"), py::arg("down"), py::arg("parts"),
           py::arg("defer_ack") = false)
      .def("wait_ack", &XferClient::wait_ack, py::arg("handle"),
           py::arg("timeout_s") = 60.0,
           py::call_guard<py::gil_scoped_release>())
      .def("close_all", &XferClient::close_all,
           py::call_guard<py::gil_scoped_release>());
}
