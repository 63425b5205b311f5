// Minimal HTTP/2 (RFC 7540) connection core for the native gRPC transport:
// frame IO, settings, flow control. Shared by the server and client in
// grpc_transport.cpp.
//
// Why this exists: the round-1 ceiling attribution (profiles/README.md)
// showed the python-grpcio stack caps the 19 MB Predict loopback at
// ~9.5 GB/s of copy bandwidth (~12 buffer traversals per hop). This core
// speaks standard gRPC-over-HTTP/2 (interop-tested against grpcio in both
// directions) with ~2 copies per hop: sends gather straight from the wire
// buffer via writev (the MI355X-native analogue of the reference's
// two-slice zero-copy encode, grpc_tensor_coding.cc:140-248), and receives
// read DATA payloads directly into the message buffer.
#pragma once

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <sys/uio.h>
#include <sys/un.h>
#include <unistd.h>

#include <cerrno>

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <mutex>
#include <new>
#include <stdexcept>
#include <string>
#include <unordered_map>
#include <vector>

namespace h2 {

// ---------------------------------------------------------------------------
// errors
// ---------------------------------------------------------------------------
struct ConnError : std::runtime_error {
  using std::runtime_error::runtime_error;
};

// ---------------------------------------------------------------------------
// frame constants
// ---------------------------------------------------------------------------
enum FrameType : uint8_t {
  F_DATA = 0, F_HEADERS = 1, F_PRIORITY = 2, F_RST_STREAM = 3,
  F_SETTINGS = 4, F_PUSH_PROMISE = 5, F_PING = 6, F_GOAWAY = 7,
  F_WINDOW_UPDATE = 8, F_CONTINUATION = 9,
};
enum Flags : uint8_t {
  FL_END_STREAM = 0x1, FL_ACK = 0x1, FL_END_HEADERS = 0x4,
  FL_PADDED = 0x8, FL_PRIORITY = 0x20,
};
enum SettingsId : uint16_t {
  S_HEADER_TABLE_SIZE = 1, S_ENABLE_PUSH = 2, S_MAX_CONCURRENT_STREAMS = 3,
  S_INITIAL_WINDOW_SIZE = 4, S_MAX_FRAME_SIZE = 5, S_MAX_HEADER_LIST_SIZE = 6,
};

constexpr char kPreface[] = "PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n";
constexpr size_t kPrefaceLen = 24;
// what we announce: effectively-unbounded receive windows (no incremental
// stream WINDOW_UPDATEs needed for <2GB unary messages) and 16MB frames
constexpr uint32_t kOurMaxFrame = 16 * 1024 * 1024 - 1;
constexpr int64_t kOurInitialWindow = 0x7fffffff;

// ---------------------------------------------------------------------------
// malloc'd move-only buffer (avoids std::string's zero-fill on resize)
// ---------------------------------------------------------------------------
struct Buf {
  uint8_t* p = nullptr;
  size_t len = 0;
  size_t cap = 0;
  // custom deallocation (e.g. return pinned memory to a pool); null =>
  // plain free()
  void (*deleter)(void* ctx, uint8_t* p, size_t cap) = nullptr;
  void* deleter_ctx = nullptr;
  Buf() = default;
  explicit Buf(size_t n) { alloc(n); }
  Buf(const Buf&) = delete;
  Buf& operator=(const Buf&) = delete;
  Buf(Buf&& o) noexcept
      : p(o.p), len(o.len), cap(o.cap), deleter(o.deleter),
        deleter_ctx(o.deleter_ctx) {
    o.p = nullptr; o.len = 0; o.cap = 0;
    o.deleter = nullptr; o.deleter_ctx = nullptr;
  }
  Buf& operator=(Buf&& o) noexcept {
    if (this != &o) {
      release();
      p = o.p; len = o.len; cap = o.cap;
      deleter = o.deleter; deleter_ctx = o.deleter_ctx;
      o.p = nullptr; o.len = 0; o.cap = 0;
      o.deleter = nullptr; o.deleter_ctx = nullptr;
    }
    return *this;
  }
  ~Buf() { release(); }
  void release() {
    if (p != nullptr) {
      if (deleter) deleter(deleter_ctx, p, cap);
      else free(p);
    }
    p = nullptr; len = 0; cap = 0;
    deleter = nullptr; deleter_ctx = nullptr;
  }
  void alloc(size_t n) {
    release();
    p = static_cast<uint8_t*>(malloc(n ? n : 1));
    if (!p) throw std::bad_alloc();
    cap = n;
    len = 0;
  }
  // take ownership of externally allocated memory with its deleter
  void adopt(uint8_t* ptr, size_t capacity,
             void (*del)(void*, uint8_t*, size_t), void* ctx) {
    release();
    p = ptr; cap = capacity; len = 0;
    deleter = del; deleter_ctx = ctx;
  }
};

// ---------------------------------------------------------------------------
// socket io
// ---------------------------------------------------------------------------
inline void read_full(int fd, uint8_t* buf, size_t n) {
  size_t got = 0;
  while (got < n) {
    ssize_t r = ::read(fd, buf + got, n - got);
    if (r == 0) throw ConnError("connection closed by peer");
    if (r < 0) {
      if (errno == EINTR) continue;
      throw ConnError(std::string("read: ") + strerror(errno));
    }
    got += size_t(r);
  }
}

inline void discard(int fd, size_t n) {
  uint8_t scratch[4096];
  while (n > 0) {
    size_t take = n < sizeof(scratch) ? n : sizeof(scratch);
    read_full(fd, scratch, take);
    n -= take;
  }
}

inline void writev_all(int fd, struct iovec* iov, int iovcnt) {
  while (iovcnt > 0) {
    ssize_t w = ::writev(fd, iov, iovcnt);
    if (w < 0) {
      if (errno == EINTR) continue;
      throw ConnError(std::string("writev: ") + strerror(errno));
    }
    size_t left = size_t(w);
    while (iovcnt > 0 && left >= iov->iov_len) {
      left -= iov->iov_len;
      ++iov;
      --iovcnt;
    }
    if (iovcnt > 0 && left > 0) {
      iov->iov_base = static_cast<uint8_t*>(iov->iov_base) + left;
      iov->iov_len -= left;
    }
  }
}

inline void write_all(int fd, const uint8_t* buf, size_t n) {
  struct iovec iov{const_cast<uint8_t*>(buf), n};
  writev_all(fd, &iov, 1);
}

// Large socket buffers: a 19 MB message over the default ~208 KB unix
// buffer costs ~100 wakeup round-trips; 8 MB buffers cut that ~40x.
inline void tune_socket(int fd) {
  int sz = 8 * 1024 * 1024;
  ::setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &sz, sizeof(sz));
  ::setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &sz, sizeof(sz));
}

// target: "unix:///path" or "unix:/path" or "host:port"
inline int connect_target(const std::string& target) {
  int fd;
  if (target.rfind("unix:", 0) == 0) {
    std::string path = target.substr(5);
    while (path.size() >= 2 && path[0] == '/' && path[1] == '/')
      path = path.substr(1);  // unix:///p -> /p
    fd = ::socket(AF_UNIX, SOCK_STREAM, 0);
    if (fd < 0) throw ConnError("socket: " + std::string(strerror(errno)));
    sockaddr_un addr{};
    addr.sun_family = AF_UNIX;
    if (path.size() >= sizeof(addr.sun_path))
      throw ConnError("unix path too long");
    std::memcpy(addr.sun_path, path.c_str(), path.size() + 1);
    tune_socket(fd);
    if (::connect(fd, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) < 0) {
      int e = errno;
      ::close(fd);
      throw ConnError("connect " + path + ": " + strerror(e));
    }
  } else {
    auto colon = target.rfind(':');
    if (colon == std::string::npos) throw ConnError("bad target " + target);
    std::string host = target.substr(0, colon);
    int port = std::atoi(target.c_str() + colon + 1);
    fd = ::socket(AF_INET, SOCK_STREAM, 0);
    if (fd < 0) throw ConnError("socket: " + std::string(strerror(errno)));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons(uint16_t(port));
    if (host.empty() || host == "localhost") host = "127.0.0.1";
    if (::inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1) {
      ::close(fd);
      throw ConnError("bad host " + host);
    }
    tune_socket(fd);
    if (::connect(fd, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) < 0) {
      int e = errno;
      ::close(fd);
      throw ConnError("connect " + target + ": " + strerror(e));
    }
    int one = 1;
    ::setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
  }
  return fd;
}

// ---------------------------------------------------------------------------
// frame header
// ---------------------------------------------------------------------------
struct FrameHeader {
  uint32_t length;
  uint8_t type;
  uint8_t flags;
  uint32_t stream;
};

inline void put_frame_header(uint8_t* h, uint32_t len, uint8_t type,
                             uint8_t flags, uint32_t stream) {
  h[0] = uint8_t(len >> 16);
  h[1] = uint8_t(len >> 8);
  h[2] = uint8_t(len);
  h[3] = type;
  h[4] = flags;
  h[5] = uint8_t((stream >> 24) & 0x7f);
  h[6] = uint8_t(stream >> 16);
  h[7] = uint8_t(stream >> 8);
  h[8] = uint8_t(stream);
}

inline FrameHeader read_frame_header(int fd) {
  uint8_t h[9];
  read_full(fd, h, 9);
  FrameHeader fh;
  fh.length = (uint32_t(h[0]) << 16) | (uint32_t(h[1]) << 8) | h[2];
  fh.type = h[3];
  fh.flags = h[4];
  fh.stream = ((uint32_t(h[5]) & 0x7f) << 24) | (uint32_t(h[6]) << 16) |
              (uint32_t(h[7]) << 8) | h[8];
  return fh;
}

inline uint32_t be32(const uint8_t* p) {
  return (uint32_t(p[0]) << 24) | (uint32_t(p[1]) << 16) |
         (uint32_t(p[2]) << 8) | p[3];
}
inline void put_be32(uint8_t* p, uint32_t v) {
  p[0] = uint8_t(v >> 24); p[1] = uint8_t(v >> 16);
  p[2] = uint8_t(v >> 8); p[3] = uint8_t(v);
}

// ---------------------------------------------------------------------------
// connection state shared by reader + writers
// ---------------------------------------------------------------------------
struct Conn {
  int fd = -1;
  std::mutex write_mu;

  // send-side flow control (peer-advertised)
  std::mutex fc_mu;
  std::condition_variable fc_cv;
  int64_t conn_send_window = 65535;
  int64_t peer_initial_window = 65535;
  uint32_t peer_max_frame = 16384;
  int64_t peer_max_streams = -1;  // -1 = unlimited
  std::unordered_map<uint32_t, int64_t> stream_send_window;
  bool peer_settings_seen = false;
  bool broken = false;
  std::string broken_why;

  // receive-side flow control bookkeeping (we advertise kOurInitialWindow)
  uint64_t recv_consumed_since_update = 0;

  explicit Conn(int fd_) : fd(fd_) {}
  ~Conn() {
    if (fd >= 0) ::close(fd);
  }

  void mark_broken(const std::string& why) {
    {
      std::lock_guard<std::mutex> lk(fc_mu);
      if (!broken) {
        broken = true;
        broken_why = why;
      }
    }
    fc_cv.notify_all();
    ::shutdown(fd, SHUT_RDWR);
  }

  // -- handshake helpers ----------------------------------------------------
  // our SETTINGS + a connection WINDOW_UPDATE raising the 64KB default to
  // kOurInitialWindow (one frame each; batched into one write)
  void send_initial_settings() {
    uint8_t out[9 + 18 + 9 + 4];
    uint8_t* p = out;
    put_frame_header(p, 18, F_SETTINGS, 0, 0);
    p += 9;
    auto setting = [&](uint16_t id, uint32_t val) {
      p[0] = uint8_t(id >> 8); p[1] = uint8_t(id);
      put_be32(p + 2, val);
      p += 6;
    };
    setting(S_INITIAL_WINDOW_SIZE, uint32_t(kOurInitialWindow));
    setting(S_MAX_FRAME_SIZE, kOurMaxFrame);
    setting(S_MAX_HEADER_LIST_SIZE, 1 << 20);
    put_frame_header(p, 4, F_WINDOW_UPDATE, 0, 0);
    p += 9;
    put_be32(p, uint32_t(kOurInitialWindow - 65535));
    p += 4;
    std::lock_guard<std::mutex> lk(write_mu);
    write_all(fd, out, size_t(p - out));
  }

  void send_settings_ack() {
    uint8_t out[9];
    put_frame_header(out, 0, F_SETTINGS, FL_ACK, 0);
    std::lock_guard<std::mutex> lk(write_mu);
    write_all(fd, out, 9);
  }

  void send_ping_ack(const uint8_t* opaque) {
    uint8_t out[9 + 8];
    put_frame_header(out, 8, F_PING, FL_ACK, 0);
    std::memcpy(out + 9, opaque, 8);
    std::lock_guard<std::mutex> lk(write_mu);
    write_all(fd, out, 17);
  }

  void send_rst_stream(uint32_t stream, uint32_t code) {
    uint8_t out[9 + 4];
    put_frame_header(out, 4, F_RST_STREAM, 0, stream);
    put_be32(out + 9, code);
    std::lock_guard<std::mutex> lk(write_mu);
    write_all(fd, out, 13);
  }

  // -- receive-side window refill -------------------------------------------
  // called by the reader after consuming a DATA frame's bytes
  void account_received(uint64_t n) {
    recv_consumed_since_update += n;
    if (recv_consumed_since_update >= (uint64_t(1) << 29)) {
      uint8_t out[9 + 4];
      put_frame_header(out, 4, F_WINDOW_UPDATE, 0, 0);
      put_be32(out + 9, uint32_t(recv_consumed_since_update));
      recv_consumed_since_update = 0;
      std::lock_guard<std::mutex> lk(write_mu);
      write_all(fd, out, 13);
    }
  }

  // -- peer frame processing shared by client/server readers ---------------
  void apply_peer_settings(const uint8_t* payload, uint32_t len) {
    std::lock_guard<std::mutex> lk(fc_mu);
    for (uint32_t off = 0; off + 6 <= len; off += 6) {
      uint16_t id = uint16_t((payload[off] << 8) | payload[off + 1]);
      uint32_t val = be32(payload + off + 2);
      switch (id) {
        case S_INITIAL_WINDOW_SIZE: {
          int64_t delta = int64_t(val) - peer_initial_window;
          peer_initial_window = int64_t(val);
          for (auto& kv : stream_send_window) kv.second += delta;
          break;
        }
        case S_MAX_FRAME_SIZE:
          peer_max_frame = val;
          break;
        case S_MAX_CONCURRENT_STREAMS:
          peer_max_streams = int64_t(val);
          break;
        default:
          break;  // header table size ignored: our encoder is stateless
      }
    }
    peer_settings_seen = true;
    fc_cv.notify_all();
  }

  void apply_window_update(uint32_t stream, uint32_t increment) {
    {
      std::lock_guard<std::mutex> lk(fc_mu);
      if (stream == 0) {
        conn_send_window += increment;
      } else {
        auto it = stream_send_window.find(stream);
        if (it != stream_send_window.end()) it->second += increment;
      }
    }
    fc_cv.notify_all();
  }

  void open_send_stream(uint32_t stream) {
    std::lock_guard<std::mutex> lk(fc_mu);
    stream_send_window[stream] = peer_initial_window;
  }

  void close_send_stream(uint32_t stream) {
    std::lock_guard<std::mutex> lk(fc_mu);
    stream_send_window.erase(stream);
  }

  // Sends one gRPC message as flow-controlled DATA frames gathered from
  // (prefix5 + payload) with NO copy of the payload: every frame is a
  // writev of [9B header, payload slices]. end_stream set on the last
  // frame when `end_stream` (client request); servers follow with trailers.
  void send_data_message(uint32_t stream, const uint8_t* payload, size_t n,
                         bool end_stream) {
    uint8_t prefix[5];
    prefix[0] = 0;  // not compressed
    put_be32(prefix + 1, uint32_t(n));
    size_t total = n + 5;
    size_t sent = 0;  // offset into virtual (prefix || payload)
    while (sent < total) {
      size_t chunk;
      {
        std::unique_lock<std::mutex> lk(fc_mu);
        fc_cv.wait(lk, [&] {
          if (broken) return true;
          auto it = stream_send_window.find(stream);
          int64_t sw = it == stream_send_window.end() ? 0 : it->second;
          return conn_send_window > 0 && sw > 0;
        });
        if (broken) throw ConnError("connection broken: " + broken_why);
        int64_t sw = stream_send_window[stream];
        int64_t avail = conn_send_window < sw ? conn_send_window : sw;
        chunk = size_t(avail);
        if (chunk > total - sent) chunk = total - sent;
        if (chunk > peer_max_frame) chunk = peer_max_frame;
        conn_send_window -= int64_t(chunk);
        stream_send_window[stream] -= int64_t(chunk);
      }
      bool last = (sent + chunk == total);
      uint8_t fh[9];
      put_frame_header(fh, uint32_t(chunk), F_DATA,
                       (last && end_stream) ? FL_END_STREAM : 0, stream);
      struct iovec iov[3];
      int iovcnt = 0;
      iov[iovcnt++] = {fh, 9};
      size_t off = sent;
      size_t left = chunk;
      if (off < 5) {
        size_t pre = 5 - off < left ? 5 - off : left;
        iov[iovcnt++] = {prefix + off, pre};
        off += pre;
        left -= pre;
      }
      if (left > 0) {
        iov[iovcnt++] = {const_cast<uint8_t*>(payload) + (off - 5), left};
      }
      {
        std::lock_guard<std::mutex> lk(write_mu);
        writev_all(fd, iov, iovcnt);
      }
      sent += chunk;
    }
  }

  // header block in one HEADERS frame (our blocks are tiny)
  void send_headers(uint32_t stream, const std::string& block,
                    bool end_stream) {
    uint8_t fh[9];
    put_frame_header(fh, uint32_t(block.size()), F_HEADERS,
                     uint8_t(FL_END_HEADERS |
                             (end_stream ? FL_END_STREAM : 0)),
                     stream);
    struct iovec iov[2] = {
        {fh, 9},
        {const_cast<char*>(block.data()), block.size()},
    };
    std::lock_guard<std::mutex> lk(write_mu);
    writev_all(fd, iov, 2);
  }
};

// Streams ONE gRPC message of known total payload length as flow-controlled
// DATA frames, fed incrementally: write() is called with consecutive byte
// spans that must sum to exactly payload_len. This lets a producer
// interleave generation with transmission — the transport's streaming send
// hands each pinned-staging DMA chunk here while the NEXT chunk is still
// copying off the device (the north star's "hipMemcpyAsync overlaps the
// gRPC send"). END_STREAM is set on the frame that completes the message
// iff `end_stream`.
struct DataMessageWriter {
  Conn& conn;
  uint32_t stream;
  size_t total;  // 5-byte gRPC prefix + payload
  bool end_stream;
  uint8_t prefix[5];
  size_t sent = 0;  // virtual offset into (prefix || payload)

  DataMessageWriter(Conn& c, uint32_t s, size_t payload_len, bool es)
      : conn(c), stream(s), total(payload_len + 5), end_stream(es) {
    prefix[0] = 0;  // not compressed
    put_be32(prefix + 1, uint32_t(payload_len));
  }

  // feed the next `n` payload bytes (in order); sends them (plus the
  // prefix, on the first call) as DATA frames respecting both windows
  void write(const uint8_t* payload, size_t n) {
    size_t virt_avail = (sent < 5 ? 5 - sent : 0) + n;
    size_t consumed_payload = 0;
    while (virt_avail > 0) {
      size_t chunk;
      {
        std::unique_lock<std::mutex> lk(conn.fc_mu);
        conn.fc_cv.wait(lk, [&] {
          if (conn.broken) return true;
          auto it = conn.stream_send_window.find(stream);
          int64_t sw = it == conn.stream_send_window.end() ? 0 : it->second;
          return conn.conn_send_window > 0 && sw > 0;
        });
        if (conn.broken)
          throw ConnError("connection broken: " + conn.broken_why);
        int64_t sw = conn.stream_send_window[stream];
        int64_t avail =
            conn.conn_send_window < sw ? conn.conn_send_window : sw;
        chunk = size_t(avail);
        if (chunk > virt_avail) chunk = virt_avail;
        if (chunk > conn.peer_max_frame) chunk = conn.peer_max_frame;
        conn.conn_send_window -= int64_t(chunk);
        conn.stream_send_window[stream] -= int64_t(chunk);
      }
      bool last = (sent + chunk == total);
      uint8_t fh[9];
      put_frame_header(fh, uint32_t(chunk), F_DATA,
                       (last && end_stream) ? FL_END_STREAM : 0, stream);
      struct iovec iov[3];
      int iovcnt = 0;
      iov[iovcnt++] = {fh, 9};
      size_t left = chunk;
      if (sent < 5) {
        size_t pre = 5 - sent < left ? 5 - sent : left;
        iov[iovcnt++] = {prefix + sent, pre};
        left -= pre;
      }
      if (left > 0) {
        iov[iovcnt++] = {const_cast<uint8_t*>(payload) + consumed_payload,
                         left};
      }
      {
        std::lock_guard<std::mutex> lk(conn.write_mu);
        writev_all(conn.fd, iov, iovcnt);
      }
      sent += chunk;
      consumed_payload += left;
      virt_avail -= chunk;
    }
  }

  // flush a zero-payload message's prefix (write() never called)
  void finish() {
    if (sent < total && total == 5) write(nullptr, 0);
  }

  bool complete() const { return sent == total; }
};

// grpc-message percent coding (gRPC HTTP/2 protocol spec)
inline std::string percent_encode(const std::string& s) {
  static const char* hex = "0123456789ABCDEF";
  std::string out;
  out.reserve(s.size());
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

inline std::string percent_decode(const std::string& s) {
  std::string out;
  out.reserve(s.size());
  for (size_t i = 0; i < s.size(); ++i) {
    if (s[i] == '%' && i + 2 < s.size()) {
      auto hexval = [](char c) -> int {
        if (c >= '0' && c <= '9') return c - '0';
        if (c >= 'a' && c <= 'f') return c - 'a' + 10;
        if (c >= 'A' && c <= 'F') return c - 'A' + 10;
        return -1;
      };
      int hi = hexval(s[i + 1]), lo = hexval(s[i + 2]);
      if (hi >= 0 && lo >= 0) {
        out.push_back(char((hi << 4) | lo));
        i += 2;
        continue;
      }
    }
    out.push_back(s[i]);
  }
  return out;
}

}  // namespace h2
