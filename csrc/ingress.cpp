// Native gRPC ingress daemon (reference: TF-Serving's
// C++ grpc server front-end).
//
// grpc-python's server tops out at ~28 msgs/s of 3 MB on this stack
// (profiles/r02 serving table) — the GIL-bound ingress, not the model,
// is the full-stack ceiling. This daemon owns the sockets and the
// HTTP/2 state machine in C++ threads (no GIL), and only enters Python
// for the actual Predict dispatch: one callback per complete request
// carrying (path, payload bytes), returning (grpc_status, message,
// response bytes). Protobuf stays in serving/wireproto.py.
//
// HTTP/2 (RFC 7540) is implemented directly: frame parsing, SETTINGS,
// PING, flow control both directions, CONTINUATION, padded/priority
// flags. HPACK (RFC 7541) header *decompression* is delegated to the
// system libnghttp2 (linked -l:libnghttp2.so.14; the image has no
// nghttp2 headers, so the small stable inflater API is declared below).
// Response headers are HPACK-encoded by hand as literals without
// indexing (always legal, no Huffman required).
//
// Scope: insecure (h2c prior-knowledge, as used by grpc insecure
// channels), unary calls only — exactly the PredictionService surface.

#include <pybind11/pybind11.h>

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <algorithm>
#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <memory>
#include <mutex>
#include <string>
#include <thread>
#include <unordered_map>
#include <unordered_set>
#include <vector>

namespace py = pybind11;

// ---------------------------------------------------------------- nghttp2
// Minimal declarations of libnghttp2.so.14's HPACK inflater API (the
// /usr/include headers are absent from this image; these signatures are
// the stable public API of nghttp2 >= 1.x).
extern "C" {
typedef struct nghttp2_hd_inflater nghttp2_hd_inflater;
typedef struct {
  uint8_t* name;
  uint8_t* value;
  size_t namelen;
  size_t valuelen;
  uint8_t flags;
} nghttp2_nv;
int nghttp2_hd_inflate_new(nghttp2_hd_inflater** inflater_ptr);
void nghttp2_hd_inflate_del(nghttp2_hd_inflater* inflater);
ssize_t nghttp2_hd_inflate_hd2(nghttp2_hd_inflater* inflater,
                               nghttp2_nv* nv_out, int* inflate_flags,
                               const uint8_t* in, size_t inlen,
                               int in_final);
int nghttp2_hd_inflate_end_headers(nghttp2_hd_inflater* inflater);
}
enum { HD_INFLATE_FINAL = 0x01, HD_INFLATE_EMIT = 0x02 };

namespace {

// ------------------------------------------------------------- h2 consts
enum FrameType : uint8_t {
  F_DATA = 0x0,
  F_HEADERS = 0x1,
  F_PRIORITY = 0x2,
  F_RST_STREAM = 0x3,
  F_SETTINGS = 0x4,
  F_PUSH_PROMISE = 0x5,
  F_PING = 0x6,
  F_GOAWAY = 0x7,
  F_WINDOW_UPDATE = 0x8,
  F_CONTINUATION = 0x9,
};
enum Flags : uint8_t {
  FL_END_STREAM = 0x1,  // also SETTINGS/PING ACK
  FL_END_HEADERS = 0x4,
  FL_PADDED = 0x8,
  FL_PRIORITY = 0x20,
};
constexpr uint32_t OUR_MAX_FRAME = 4u << 20;       // we accept 4 MB frames
constexpr uint32_t OUR_CONN_WINDOW = 0x7fffffffu;  // ~2 GB recv window
constexpr uint32_t OUR_STREAM_WINDOW = 0x40000000u;
constexpr size_t MAX_BODY = 256u << 20;  // grpc.max_receive_message_length
constexpr uint32_t MAX_STREAMS = 128;    // advertised + enforced per conn
const char PREFACE[] = "PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n";

inline uint32_t rd_u32(const uint8_t* p) {
  return (uint32_t(p[0]) << 24) | (uint32_t(p[1]) << 16) |
         (uint32_t(p[2]) << 8) | p[3];
}
inline void wr_u32(uint8_t* p, uint32_t v) {
  p[0] = v >> 24;
  p[1] = v >> 16;
  p[2] = v >> 8;
  p[3] = v;
}
inline void frame_header(uint8_t* p, uint32_t len, uint8_t type,
                         uint8_t flags, uint32_t sid) {
  p[0] = len >> 16;
  p[1] = len >> 8;
  p[2] = len;
  p[3] = type;
  p[4] = flags;
  wr_u32(p + 5, sid & 0x7fffffffu);
}

// HPACK 7-bit-prefix integer (RFC 7541 §5.1) — string lengths can
// exceed 126 (long grpc-message error details), so the multi-byte
// form is required, not optional.
void hpack_len7(std::vector<uint8_t>& out, size_t v) {
  if (v < 127) {
    out.push_back(uint8_t(v));
    return;
  }
  out.push_back(0x7f);
  v -= 127;
  while (v >= 128) {
    out.push_back(uint8_t((v & 0x7f) | 0x80));
    v >>= 7;
  }
  out.push_back(uint8_t(v));
}

// HPACK literal-without-indexing, new name, no Huffman (RFC 7541 §6.2.2).
void hpack_literal(std::vector<uint8_t>& out, const std::string& name,
                   const std::string& value) {
  out.push_back(0x00);
  hpack_len7(out, name.size());
  out.insert(out.end(), name.begin(), name.end());
  hpack_len7(out, value.size());
  out.insert(out.end(), value.begin(), value.end());
}

struct Stream {
  std::string path;
  std::vector<uint8_t> body;
  std::vector<uint8_t> header_block;  // fragments until END_HEADERS
  bool headers_done = false;
  bool end_stream = false;
  bool refused = false;  // over MAX_STREAMS: RST after HPACK decode
  int64_t send_window = 65535;  // reset from peer INITIAL_WINDOW_SIZE
  // Absolute deadline from the client's grpc-timeout header (steady
  // clock); time_point::max() = none.
  std::chrono::steady_clock::time_point deadline =
      std::chrono::steady_clock::time_point::max();
};

// grpc-timeout: "<digits><unit>", unit in H M S m u n (RFC: gRPC
// HTTP/2 transport spec). Returns max() on parse failure.
inline std::chrono::steady_clock::time_point
parse_grpc_timeout(const uint8_t* v, size_t n) {
  if (n < 2 || n > 9) return std::chrono::steady_clock::time_point::max();
  uint64_t val = 0;
  for (size_t i = 0; i + 1 < n; ++i) {
    if (v[i] < '0' || v[i] > '9')
      return std::chrono::steady_clock::time_point::max();
    val = val * 10 + (v[i] - '0');
  }
  uint64_t ns;
  switch (v[n - 1]) {
    case 'H': ns = val * 3600ull * 1000000000ull; break;
    case 'M': ns = val * 60ull * 1000000000ull; break;
    case 'S': ns = val * 1000000000ull; break;
    case 'm': ns = val * 1000000ull; break;
    case 'u': ns = val * 1000ull; break;
    case 'n': ns = val; break;
    default: return std::chrono::steady_clock::time_point::max();
  }
  return std::chrono::steady_clock::now() +
         std::chrono::nanoseconds(ns);
}

struct Conn {
  int fd = -1;
  std::mutex wmu;  // serializes writes
  std::mutex mu;   // protects below
  std::condition_variable cv;
  int64_t conn_send_window = 65535;
  uint32_t peer_max_frame = 16384;
  uint32_t peer_initial_window = 65535;
  std::unordered_map<uint32_t, std::shared_ptr<Stream>> streams;
  std::atomic<bool> dead{false};
  nghttp2_hd_inflater* inflater = nullptr;
  uint64_t unacked = 0;  // received DATA not yet WINDOW_UPDATEd

  ~Conn() {
    if (inflater) nghttp2_hd_inflate_del(inflater);
    if (fd >= 0) ::close(fd);
  }
  bool write_all(const uint8_t* p, size_t n) {
    std::lock_guard<std::mutex> lk(wmu);
    while (n) {
      ssize_t w = ::send(fd, p, n, MSG_NOSIGNAL);
      if (w <= 0) {
        dead = true;
        cv.notify_all();
        return false;
      }
      p += w;
      n -= size_t(w);
    }
    return true;
  }
  bool write_frame(uint8_t type, uint8_t flags, uint32_t sid,
                   const uint8_t* payload, size_t n) {
    std::vector<uint8_t> buf(9 + n);
    frame_header(buf.data(), uint32_t(n), type, flags, sid);
    if (n) memcpy(buf.data() + 9, payload, n);
    return write_all(buf.data(), buf.size());
  }
};

struct Request {
  std::shared_ptr<Conn> conn;
  uint32_t sid;
  std::shared_ptr<Stream> stream;  // kept live for its send_window
  std::string path;
  std::vector<uint8_t> body;  // raw h2 DATA (gRPC length-prefixed)
};

// ------------------------------------------------------------- the server
class IngressServer {
 public:
  IngressServer(int port, py::function handler, int workers)
      : handler_(std::move(handler)),
        n_workers_(std::max(1, workers)) {
    listen_fd_ = ::socket(AF_INET6, SOCK_STREAM, 0);
    bool v6 = listen_fd_ >= 0;
    if (!v6) listen_fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
    if (listen_fd_ < 0) throw std::runtime_error("socket() failed");
    int one = 1;
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    if (v6) {
      int zero = 0;  // dual-stack
      setsockopt(listen_fd_, IPPROTO_IPV6, IPV6_V6ONLY, &zero,
                 sizeof(zero));
      sockaddr_in6 a{};
      a.sin6_family = AF_INET6;
      a.sin6_addr = in6addr_any;
      a.sin6_port = htons(uint16_t(port));
      if (::bind(listen_fd_, (sockaddr*)&a, sizeof(a)) != 0) {
        ::close(listen_fd_);
        throw std::runtime_error("bind() failed");
      }
      socklen_t len = sizeof(a);
      getsockname(listen_fd_, (sockaddr*)&a, &len);
      port_ = ntohs(a.sin6_port);
    } else {
      sockaddr_in a{};
      a.sin_family = AF_INET;
      a.sin_addr.s_addr = INADDR_ANY;
      a.sin_port = htons(uint16_t(port));
      if (::bind(listen_fd_, (sockaddr*)&a, sizeof(a)) != 0) {
        ::close(listen_fd_);
        throw std::runtime_error("bind() failed");
      }
      socklen_t len = sizeof(a);
      getsockname(listen_fd_, (sockaddr*)&a, &len);
      port_ = ntohs(a.sin_port);
    }
    if (::listen(listen_fd_, 128) != 0) {
      ::close(listen_fd_);
      throw std::runtime_error("listen() failed");
    }
  }

  ~IngressServer() { stop(); }

  int port() const { return port_; }

  void start() {
    if (running_.exchange(true)) return;
    accept_thread_ = std::thread([this] { accept_loop(); });
    for (int i = 0; i < n_workers_; ++i)
      workers_.emplace_back([this] { worker_loop(); });
  }

  void stop(double drain_s = 1.0) {
    if (!running_.exchange(false)) return;
    ::shutdown(listen_fd_, SHUT_RDWR);  // no new connections
    {
      std::lock_guard<std::mutex> lk(qmu_);
      qcv_.notify_all();
    }
    {  // Drain: give queued + in-flight requests up to drain_s to
       // finish and their responses to flush BEFORE killing the
       // connections (TF-Serving's graceful-shutdown behavior).
      py::gil_scoped_release rel;  // workers need the GIL to finish
      auto deadline = std::chrono::steady_clock::now() +
          std::chrono::duration<double>(drain_s);
      while (std::chrono::steady_clock::now() < deadline) {
        bool idle;
        {
          std::lock_guard<std::mutex> lk(qmu_);
          idle = queue_.empty() && in_flight_.load() == 0;
        }
        if (idle) break;
        std::this_thread::sleep_for(std::chrono::milliseconds(10));
      }
    }
    {  // GOAWAY(NO_ERROR) then kick every live connection so reader
       // threads return; grpc clients surface UNAVAILABLE instead of a
       // raw socket reset.
      std::lock_guard<std::mutex> lk(cmu_);
      for (auto& w : conns_)
        if (auto c = w.lock()) {
          uint8_t ga[17];
          frame_header(ga, 8, F_GOAWAY, 0, 0);
          wr_u32(ga + 9, 0x7fffffffu);  // last-stream-id
          wr_u32(ga + 13, 0);           // NO_ERROR
          c->write_all(ga, 17);
          c->dead = true;
          ::shutdown(c->fd, SHUT_RDWR);
          c->cv.notify_all();
        }
    }
    py::gil_scoped_release rel;  // joined threads may need the GIL
    if (accept_thread_.joinable()) accept_thread_.join();
    for (auto& t : workers_)
      if (t.joinable()) t.join();
    workers_.clear();
    {
      // Joining MUST happen without cmu_ held: exiting conn threads
      // take cmu_ to mark themselves finished.
      std::vector<std::thread> ts;
      {
        std::lock_guard<std::mutex> lk(cmu_);
        ts.swap(conn_threads_);
        finished_.clear();
      }
      for (auto& t : ts)
        if (t.joinable()) t.join();
    }
    ::close(listen_fd_);
    listen_fd_ = -1;
  }

 private:
  // Join connection threads that have finished (their conn_loop
  // signalled done). Without this a long-running daemon accumulates a
  // zombie std::thread handle + expired weak_ptr per client that ever
  // connected. Called with cmu_ HELD.
  void reap_locked() {
    if (finished_.empty()) return;
    for (auto it = conn_threads_.begin(); it != conn_threads_.end();) {
      auto fin = finished_.find(it->get_id());
      if (fin != finished_.end()) {
        it->join();
        finished_.erase(fin);
        it = conn_threads_.erase(it);
      } else {
        ++it;
      }
    }
    conns_.erase(std::remove_if(conns_.begin(), conns_.end(),
                                [](const std::weak_ptr<Conn>& w) {
                                  return w.expired();
                                }),
                 conns_.end());
  }

  // ------------------------------------------------------------ accept
  void accept_loop() {
    while (running_) {
      int fd = ::accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) break;
      int one = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      auto conn = std::make_shared<Conn>();
      conn->fd = fd;
      nghttp2_hd_inflate_new(&conn->inflater);
      std::lock_guard<std::mutex> lk(cmu_);
      reap_locked();
      conns_.push_back(conn);
      conn_threads_.emplace_back([this, conn] {
        conn_loop(conn);
        std::lock_guard<std::mutex> lk2(cmu_);
        finished_.insert(std::this_thread::get_id());
      });
    }
  }

  // ------------------------------------------------------- reader logic
  struct Reader {
    int fd;
    std::vector<uint8_t> buf;
    size_t pos = 0;
    bool need(size_t n) {
      while (buf.size() - pos < n) {
        if (pos > (1u << 20)) {  // compact
          buf.erase(buf.begin(), buf.begin() + pos);
          pos = 0;
        }
        uint8_t tmp[65536];
        ssize_t r = ::recv(fd, tmp, sizeof(tmp), 0);
        if (r <= 0) return false;
        buf.insert(buf.end(), tmp, tmp + r);
      }
      return true;
    }
    const uint8_t* take(size_t n) {
      const uint8_t* p = buf.data() + pos;
      pos += n;
      return p;
    }
  };

  void conn_loop(std::shared_ptr<Conn> conn) {
    Reader rd{conn->fd, {}, 0};
    // client preface
    if (!rd.need(24) ||
        memcmp(rd.take(24), PREFACE, 24) != 0)
      return;
    // our SETTINGS + a big connection window
    {
      uint8_t s[9 + 18 + 13];
      // MAX_FRAME_SIZE(0x5), INITIAL_WINDOW_SIZE(0x4),
      // MAX_CONCURRENT_STREAMS(0x3) — bounded partial-body state.
      frame_header(s, 18, F_SETTINGS, 0, 0);
      s[9] = 0;
      s[10] = 0x5;
      wr_u32(s + 11, OUR_MAX_FRAME);
      s[15] = 0;
      s[16] = 0x4;
      wr_u32(s + 17, OUR_STREAM_WINDOW);
      s[21] = 0;
      s[22] = 0x3;
      wr_u32(s + 23, MAX_STREAMS);
      frame_header(s + 27, 4, F_WINDOW_UPDATE, 0, 0);
      wr_u32(s + 36, OUR_CONN_WINDOW - 65535);
      if (!conn->write_all(s, sizeof(s))) return;
    }
    uint32_t continuation_sid = 0;  // nonzero while expecting CONTINUATION
    while (!conn->dead && running_) {
      if (!rd.need(9)) break;
      const uint8_t* h = rd.take(9);
      uint32_t len = (uint32_t(h[0]) << 16) | (uint32_t(h[1]) << 8) | h[2];
      uint8_t type = h[3], flags = h[4];
      uint32_t sid = rd_u32(h + 5) & 0x7fffffffu;
      if (len > OUR_MAX_FRAME + 1024) break;  // protocol error
      if (!rd.need(len)) break;
      const uint8_t* p = rd.take(len);
      if (continuation_sid && (type != F_CONTINUATION ||
                               sid != continuation_sid))
        break;  // RFC 7540 §6.10: CONTINUATION must follow directly
      switch (type) {
        case F_SETTINGS:
          if (flags & FL_END_STREAM) break;  // ACK of ours
          handle_settings(conn, p, len);
          conn->write_frame(F_SETTINGS, FL_END_STREAM, 0, nullptr, 0);
          break;
        case F_PING:
          if (!(flags & FL_END_STREAM) && len == 8)
            conn->write_frame(F_PING, FL_END_STREAM, 0, p, len);
          break;
        case F_WINDOW_UPDATE: {
          if (len != 4) break;
          uint32_t inc = rd_u32(p) & 0x7fffffffu;
          if (!inc) break;
          std::lock_guard<std::mutex> lk(conn->mu);
          if (sid == 0) {
            conn->conn_send_window += inc;
          } else {
            auto it = conn->streams.find(sid);
            if (it != conn->streams.end())
              it->second->send_window += inc;
          }
          conn->cv.notify_all();
          break;
        }
        case F_HEADERS: {
          if (sid == 0) return;
          size_t off = 0, end = len;
          if (flags & FL_PADDED) {
            if (!len) return;
            uint8_t pad = p[0];
            off = 1;
            if (pad + 1u > len) return;
            end = len - pad;
          }
          if (flags & FL_PRIORITY) {
            if (off + 5 > end) return;
            off += 5;
          }
          std::shared_ptr<Stream> st;
          {
            std::lock_guard<std::mutex> lk(conn->mu);
            auto it = conn->streams.find(sid);
            if (it != conn->streams.end() && it->second->headers_done)
              st = it->second;  // trailing HEADERS on an open stream
          }
          if (st) {
            // Request trailers: MUST still run through the HPACK
            // inflater (the dynamic table is connection-wide state;
            // skipping a header block would desync every later
            // request), then half-close without clobbering the body.
            // Trailers carry no :path, so finish_headers leaves
            // st->path intact and dispatches on end_stream as usual.
            st->header_block.assign(p + off, p + end);
            st->end_stream |= bool(flags & FL_END_STREAM);
            if (flags & FL_END_HEADERS) {
              if (!finish_headers(conn, sid, st)) return;
            } else {
              continuation_sid = sid;
            }
            break;
          }
          st = std::make_shared<Stream>();
          st->send_window = conn->peer_initial_window;
          st->end_stream = flags & FL_END_STREAM;
          st->header_block.assign(p + off, p + end);
          {
            std::lock_guard<std::mutex> lk(conn->mu);
            // Enforce (with slack for streams in response flight)
            // what our SETTINGS advertised: a protocol-violating
            // client does not get unbounded partial-body state. The
            // refused stream's header block must STILL run through
            // the inflater (connection-wide HPACK table), so the
            // refusal is applied after decode in finish_headers.
            st->refused = conn->streams.size() >= 2 * MAX_STREAMS;
            conn->streams[sid] = st;
          }
          if (flags & FL_END_HEADERS) {
            if (!finish_headers(conn, sid, st)) return;
          } else {
            continuation_sid = sid;
          }
          break;
        }
        case F_CONTINUATION: {
          std::shared_ptr<Stream> st;
          {
            std::lock_guard<std::mutex> lk(conn->mu);
            auto it = conn->streams.find(sid);
            if (it == conn->streams.end()) return;
            st = it->second;
          }
          st->header_block.insert(st->header_block.end(), p, p + len);
          if (flags & FL_END_HEADERS) {
            continuation_sid = 0;
            if (!finish_headers(conn, sid, st)) return;
          }
          break;
        }
        case F_DATA: {
          std::shared_ptr<Stream> st;
          {
            std::lock_guard<std::mutex> lk(conn->mu);
            auto it = conn->streams.find(sid);
            if (it != conn->streams.end()) st = it->second;
          }
          size_t off = 0, end = len;
          if (flags & FL_PADDED) {
            if (!len) return;
            uint8_t pad = p[0];
            off = 1;
            if (pad + 1u > len) return;
            end = len - pad;
          }
          if (st) {
            if (st->body.size() + (end - off) > MAX_BODY) {
              rst_stream(conn, sid, 0xb /*ENHANCE_YOUR_CALM*/);
              std::lock_guard<std::mutex> lk(conn->mu);
              conn->streams.erase(sid);
              break;
            }
            st->body.insert(st->body.end(), p + off, p + end);
          }
          // Replenish the connection window (stream windows are huge),
          // COALESCED: one tiny WINDOW_UPDATE write per ~1 MB or per
          // request end, not one per 16 KB client DATA frame — the
          // per-frame version cost ~200 extra syscalls per 3 MB
          // message and lost to grpc-python's C-core on throughput.
          conn->unacked += len;
          const bool eos = st && (flags & FL_END_STREAM);
          if (conn->unacked && (eos || conn->unacked >= (1u << 20))) {
            uint8_t wu[13];
            frame_header(wu, 4, F_WINDOW_UPDATE, 0, 0);
            wr_u32(wu + 9, uint32_t(conn->unacked));
            conn->unacked = 0;
            conn->write_all(wu, 13);
          }
          if (eos) dispatch(conn, sid, st);
          break;
        }
        case F_RST_STREAM: {
          std::lock_guard<std::mutex> lk(conn->mu);
          conn->streams.erase(sid);
          break;
        }
        case F_GOAWAY:
          return;
        case F_PRIORITY:
        case F_PUSH_PROMISE:
        default:
          break;  // ignore
      }
    }
  }

  void handle_settings(const std::shared_ptr<Conn>& conn,
                       const uint8_t* p, uint32_t len) {
    std::lock_guard<std::mutex> lk(conn->mu);
    for (uint32_t off = 0; off + 6 <= len; off += 6) {
      uint16_t id = (uint16_t(p[off]) << 8) | p[off + 1];
      uint32_t val = rd_u32(p + off + 2);
      if (id == 0x5) {
        if (val >= 16384 && val <= 16777215) conn->peer_max_frame = val;
      } else if (id == 0x4) {
        int64_t delta =
            int64_t(val) - int64_t(conn->peer_initial_window);
        conn->peer_initial_window = val;
        for (auto& kv : conn->streams)
          kv.second->send_window += delta;
        conn->cv.notify_all();
      }
    }
  }

  bool finish_headers(const std::shared_ptr<Conn>& conn, uint32_t sid,
                      const std::shared_ptr<Stream>& st) {
    const uint8_t* in = st->header_block.data();
    size_t inlen = st->header_block.size();
    for (;;) {
      nghttp2_nv nv;
      int iflags = 0;
      ssize_t rv = nghttp2_hd_inflate_hd2(conn->inflater, &nv, &iflags,
                                          in, inlen, 1);
      if (rv < 0) return false;  // HPACK error: kill the connection
      in += rv;
      inlen -= size_t(rv);
      if (iflags & HD_INFLATE_EMIT) {
        if (nv.namelen == 5 && memcmp(nv.name, ":path", 5) == 0)
          st->path.assign((const char*)nv.value, nv.valuelen);
        else if (nv.namelen == 12 &&
                 memcmp(nv.name, "grpc-timeout", 12) == 0)
          st->deadline = parse_grpc_timeout(nv.value, nv.valuelen);
      }
      if (iflags & HD_INFLATE_FINAL) {
        nghttp2_hd_inflate_end_headers(conn->inflater);
        break;
      }
      if (rv == 0 && !(iflags & (HD_INFLATE_EMIT | HD_INFLATE_FINAL)))
        return false;
    }
    st->header_block.clear();
    st->header_block.shrink_to_fit();
    st->headers_done = true;
    if (st->refused) {
      rst_stream(conn, sid, 0x7 /*REFUSED_STREAM*/);
      std::lock_guard<std::mutex> lk(conn->mu);
      conn->streams.erase(sid);
      return true;  // connection stays healthy
    }
    if (st->end_stream) dispatch(conn, sid, st);  // bodyless request
    return true;
  }

  // ------------------------------------------------------- dispatch side
  void dispatch(const std::shared_ptr<Conn>& conn, uint32_t sid,
                const std::shared_ptr<Stream>& st) {
    // The stream STAYS in conn->streams until the response is sent, so
    // client WINDOW_UPDATEs keep feeding st->send_window.
    Request r{conn, sid, st, std::move(st->path), std::move(st->body)};
    std::lock_guard<std::mutex> lk(qmu_);
    queue_.push_back(std::move(r));
    qcv_.notify_one();
  }

  void worker_loop() {
    while (true) {
      Request req;
      {
        std::unique_lock<std::mutex> lk(qmu_);
        qcv_.wait(lk, [this] { return !running_ || !queue_.empty(); });
        if (!running_ && queue_.empty()) return;
        req = std::move(queue_.front());
        queue_.pop_front();
        ++in_flight_;
      }
      serve_one(req);
      --in_flight_;
    }
  }

  void serve_one(Request& req) {
    int status = 0;
    std::string message;
    std::string resp;
    // A request whose client deadline already passed while queued must
    // not waste a model batch: answer DEADLINE_EXCEEDED immediately
    // (the client has usually RSTed the stream anyway — the write then
    // just fails quietly).
    if (req.stream &&
        std::chrono::steady_clock::now() > req.stream->deadline) {
      send_response(req.conn, req.sid, req.stream, 4,
                    "Deadline Exceeded (expired while queued)", "");
      std::lock_guard<std::mutex> lk(req.conn->mu);
      req.conn->streams.erase(req.sid);
      return;
    }
    // gRPC framing: 1-byte compressed flag + u32 length + message.
    if (req.body.size() < 5) {
      status = 13;
      message = "malformed grpc frame";
    } else if (req.body[0] != 0) {
      status = 12;  // UNIMPLEMENTED: compressed requests
      message = "compressed grpc messages not supported";
    } else {
      uint32_t mlen = rd_u32(req.body.data() + 1);
      // uint64 math: a crafted mlen near 2^32 must not wrap the
      // bounds check and read past the body.
      if (uint64_t(mlen) + 5 > req.body.size()) {
        status = 13;
        message = "truncated grpc message";
      } else {
        py::gil_scoped_acquire gil;
        try {
          py::object res = handler_(
              py::str(req.path),
              py::bytes((const char*)req.body.data() + 5, mlen));
          py::tuple t = res.cast<py::tuple>();
          status = t[0].cast<int>();
          message = t[1].cast<std::string>();
          resp = t[2].cast<std::string>();
        } catch (py::error_already_set& e) {
          status = 13;
          message = e.what();
        }
      }
    }
    req.body.clear();
    req.body.shrink_to_fit();
    send_response(req.conn, req.sid, req.stream, status, message, resp);
    std::lock_guard<std::mutex> lk(req.conn->mu);
    req.conn->streams.erase(req.sid);  // stream fully closed
  }

  static void rst_stream(const std::shared_ptr<Conn>& conn, uint32_t sid,
                         uint32_t code) {
    uint8_t f[13];
    frame_header(f, 4, F_RST_STREAM, 0, sid);
    wr_u32(f + 9, code);
    conn->write_all(f, 13);
  }

  // percent-encode for grpc-message (RFC: percent-encode non-printable)
  static std::string grpc_message_escape(const std::string& s) {
    static const char* hex = "0123456789ABCDEF";
    std::string out;
    for (unsigned char c : s) {
      if (c >= 0x20 && c <= 0x7e && c != '%') {
        out.push_back(char(c));
      } else {
        out.push_back('%');
        out.push_back(hex[c >> 4]);
        out.push_back(hex[c & 0xf]);
      }
    }
    return out;
  }

  void send_response(const std::shared_ptr<Conn>& conn, uint32_t sid,
                     const std::shared_ptr<Stream>& st, int status,
                     const std::string& message,
                     const std::string& payload) {
    if (conn->dead) return;
    // initial headers: :status 200 (static index 8) + content-type
    std::vector<uint8_t> hb;
    hb.push_back(0x88);
    hpack_literal(hb, "content-type", "application/grpc");
    if (!conn->write_frame(F_HEADERS, FL_END_HEADERS, sid, hb.data(),
                           hb.size()))
      return;
    if (status == 0) {
      // DATA: 5-byte grpc prefix + payload, chunked by peer max frame
      // and BOTH flow-control windows (connection + stream — grpc
      // clients start streams at 64 KB and grow them by WINDOW_UPDATE).
      // All frames covered by one window grant are assembled into a
      // single send() — frame-per-write was a syscall per 16 KB.
      std::vector<uint8_t> msg(5 + payload.size());
      msg[0] = 0;
      wr_u32(msg.data() + 1, uint32_t(payload.size()));
      memcpy(msg.data() + 5, payload.data(), payload.size());
      size_t off = 0;
      std::vector<uint8_t> out;
      while (off < msg.size()) {
        size_t want = msg.size() - off;
        size_t grant = 0;
        uint32_t max_frame = 16384;
        {
          std::unique_lock<std::mutex> lk(conn->mu);
          conn->cv.wait(lk, [&] {
            return conn->dead || (conn->conn_send_window > 0 &&
                                  (!st || st->send_window > 0));
          });
          if (conn->dead) return;
          grant = std::min<size_t>(
              {want, size_t(conn->conn_send_window)});
          if (st)
            grant = std::min<size_t>(grant, size_t(st->send_window));
          conn->conn_send_window -= int64_t(grant);
          if (st) st->send_window -= int64_t(grant);
          max_frame = conn->peer_max_frame;
        }
        out.clear();
        out.reserve(grant + 9 * (grant / max_frame + 1));
        size_t done = 0;
        while (done < grant) {
          size_t n = std::min<size_t>(grant - done, max_frame);
          size_t h = out.size();
          out.resize(h + 9 + n);
          frame_header(out.data() + h, uint32_t(n), F_DATA, 0, sid);
          memcpy(out.data() + h + 9, msg.data() + off + done, n);
          done += n;
        }
        if (!conn->write_all(out.data(), out.size())) return;
        off += grant;
      }
    }
    // trailers
    std::vector<uint8_t> tb;
    hpack_literal(tb, "grpc-status", std::to_string(status));
    if (!message.empty())
      hpack_literal(tb, "grpc-message", grpc_message_escape(message));
    conn->write_frame(F_HEADERS, FL_END_HEADERS | FL_END_STREAM, sid,
                      tb.data(), tb.size());
  }

  py::function handler_;
  int n_workers_;
  int listen_fd_ = -1;
  int port_ = 0;
  std::atomic<bool> running_{false};
  std::thread accept_thread_;
  std::vector<std::thread> workers_;
  std::mutex cmu_;
  std::vector<std::weak_ptr<Conn>> conns_;
  std::vector<std::thread> conn_threads_;
  std::unordered_set<std::thread::id> finished_;
  std::mutex qmu_;
  std::condition_variable qcv_;
  std::deque<Request> queue_;
  std::atomic<int> in_flight_{0};
};

}  // namespace

PYBIND11_MODULE(_ingress, m) {
  m.doc() = "native HTTP/2 gRPC ingress (nghttp2 HPACK + hand-rolled "
            "RFC 7540 framing)";
  py::class_<IngressServer>(m, "IngressServer")
      .def(py::init<int, py::function, int>(), py::arg("port"),
           py::arg("handler"), py::arg("workers") = 16)
      .def("start", &IngressServer::start,
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &IngressServer::stop, py::arg("drain_s") = 1.0)
      .def_property_readonly("port", &IngressServer::port);
}
