// Standalone unit test for the HTTP/2 connection core + HPACK codec
// (ops/csrc/h2core.h, hpack.h) — no torch, no python. Built and run under
// ASAN/UBSAN by tools/run_sanitizers.sh alongside the wire-codec test:
// these headers parse untrusted network bytes, so they must be
// memory-safe under malformed input.
#include <sys/socket.h>

#include <cassert>
#include <cstdio>
#include <cstring>
#include <random>
#include <string>
#include <thread>
#include <vector>

#include "../../min_tfs_client_amd/ops/csrc/h2core.h"
#include "../../min_tfs_client_amd/ops/csrc/hpack.h"

static int tests_run = 0;
#define CHECK(cond)                                                        \
  do {                                                                     \
    ++tests_run;                                                           \
    if (!(cond)) {                                                         \
      std::fprintf(stderr, "FAILED %s:%d: %s\n", __FILE__, __LINE__,       \
                   #cond);                                                 \
      return 1;                                                            \
    }                                                                      \
  } while (0)

// drain one gRPC message (prefix + payload) from DATA frames on `fd`,
// returning the payload; asserts frame shapes along the way
static std::vector<uint8_t> read_message(int fd, bool* saw_end_stream) {
  std::vector<uint8_t> virt;  // prefix || payload
  *saw_end_stream = false;
  while (true) {
    h2::FrameHeader fh = h2::read_frame_header(fd);
    if (fh.type != h2::F_DATA) {
      h2::discard(fd, fh.length);
      continue;
    }
    size_t off = virt.size();
    virt.resize(off + fh.length);
    h2::read_full(fd, virt.data() + off, fh.length);
    if (fh.flags & h2::FL_END_STREAM) *saw_end_stream = true;
    if (virt.size() >= 5) {
      uint32_t want = h2::be32(virt.data() + 1);
      if (virt.size() == want + 5u) break;
    }
  }
  assert(virt[0] == 0);
  return std::vector<uint8_t>(virt.begin() + 5, virt.end());
}

int test_data_message_writer_roundtrip() {
  int sv[2];
  CHECK(::socketpair(AF_UNIX, SOCK_STREAM, 0, sv) == 0);
  h2::Conn conn(sv[0]);
  conn.open_send_stream(1);
  {
    // the peer "advertised" big windows so the writer never blocks
    std::lock_guard<std::mutex> lk(conn.fc_mu);
    conn.conn_send_window = 1 << 30;
    conn.stream_send_window[1] = 1 << 30;
    conn.peer_max_frame = 4096;  // force multi-frame output
  }
  std::vector<uint8_t> payload(300000);
  for (size_t i = 0; i < payload.size(); ++i)
    payload[i] = uint8_t(i * 31 + 7);

  std::thread writer([&] {
    h2::DataMessageWriter w(conn, 1, payload.size(), true);
    // feed in awkward span sizes (1, 4095, 64K, remainder)
    size_t offs[] = {0, 1, 4096, 65536, payload.size()};
    for (int i = 0; i + 1 < 5; ++i)
      w.write(payload.data() + offs[i], offs[i + 1] - offs[i]);
    assert(w.complete());
  });
  bool end_stream = false;
  std::vector<uint8_t> got = read_message(sv[1], &end_stream);
  writer.join();
  CHECK(got == payload);
  CHECK(end_stream);
  conn.fd = -1;  // Conn dtor would close sv[0]; close both manually
  ::close(sv[0]);
  ::close(sv[1]);
  return 0;
}

int test_data_message_writer_flow_control() {
  int sv[2];
  CHECK(::socketpair(AF_UNIX, SOCK_STREAM, 0, sv) == 0);
  h2::Conn conn(sv[0]);
  conn.open_send_stream(1);
  {
    std::lock_guard<std::mutex> lk(conn.fc_mu);
    conn.conn_send_window = 1024;  // tiny: writer must wait for updates
    conn.stream_send_window[1] = 1024;
    conn.peer_max_frame = 512;
  }
  std::vector<uint8_t> payload(8192, 0xAB);
  std::thread writer([&] {
    h2::DataMessageWriter w(conn, 1, payload.size(), true);
    w.write(payload.data(), payload.size());
  });
  // reader: consume frames, drip window updates back
  std::vector<uint8_t> virt;
  bool end = false;
  while (!end) {
    h2::FrameHeader fh = h2::read_frame_header(sv[1]);
    assert(fh.type == h2::F_DATA);
    size_t off = virt.size();
    virt.resize(off + fh.length);
    h2::read_full(sv[1], virt.data() + off, fh.length);
    if (fh.flags & h2::FL_END_STREAM) end = true;
    conn.apply_window_update(0, fh.length);
    conn.apply_window_update(1, fh.length);
  }
  writer.join();
  CHECK(virt.size() == payload.size() + 5);
  CHECK(std::memcmp(virt.data() + 5, payload.data(), payload.size()) == 0);
  conn.fd = -1;
  ::close(sv[0]);
  ::close(sv[1]);
  return 0;
}

int test_hpack_roundtrip() {
  h2::HpackEncoder enc;
  std::string block;
  enc.add_indexed(&block, 3);                     // :method: POST
  enc.add_indexed(&block, 6);                     // :scheme: http
  enc.add_literal(&block, 4, "/tensorflow.serving.PredictionService/"
                             "Predict", true);    // :path, huffman
  enc.add_literal(&block, 1, "localhost");        // :authority
  enc.add_literal(&block, "te", "trailers");
  enc.add_literal(&block, 31, "application/grpc");
  h2::HpackDecoder dec;
  auto headers = dec.decode(
      reinterpret_cast<const uint8_t*>(block.data()), block.size());
  auto find = [&](const std::string& name) -> std::string {
    for (auto& h : headers)
      if (h.first == name) return h.second;
    return "<missing>";
  };
  CHECK(find(":method") == "POST");
  CHECK(find(":scheme") == "http");
  CHECK(find(":path") ==
        "/tensorflow.serving.PredictionService/Predict");
  CHECK(find(":authority") == "localhost");
  CHECK(find("te") == "trailers");
  CHECK(find("content-type") == "application/grpc");
  return 0;
}

int test_hpack_fuzz() {
  // random byte blobs must decode or throw — never crash / OOB (ASAN)
  std::mt19937 rng(424242);
  for (int iter = 0; iter < 5000; ++iter) {
    size_t n = rng() % 256;
    std::vector<uint8_t> blob(n);
    for (auto& b : blob) b = uint8_t(rng());
    h2::HpackDecoder dec;
    try {
      auto headers = dec.decode(blob.data(), blob.size());
      (void)headers;
    } catch (const std::exception&) {
      // rejected: fine
    }
  }
  ++tests_run;
  return 0;
}

int test_percent_coding() {
  std::string msg = "error: bad\nvalue \xE2\x82\xAC 100%";
  std::string enc = h2::percent_encode(msg);
  for (unsigned char c : enc)
    CHECK(c >= 0x20 && c <= 0x7e);
  CHECK(h2::percent_decode(enc) == msg);
  return 0;
}

int test_buf_adopt_deleter() {
  static int freed = 0;
  {
    h2::Buf b;
    uint8_t* raw = static_cast<uint8_t*>(malloc(64));
    b.adopt(raw, 64, [](void*, uint8_t* p, size_t) {
      free(p);
      ++freed;
    }, nullptr);
    b.len = 64;
    h2::Buf moved(std::move(b));
    CHECK(b.p == nullptr);
    CHECK(moved.cap == 64);
  }
  CHECK(freed == 1);
  return 0;
}

int main() {
  if (test_data_message_writer_roundtrip()) return 1;
  if (test_data_message_writer_flow_control()) return 1;
  if (test_hpack_roundtrip()) return 1;
  if (test_hpack_fuzz()) return 1;
  if (test_percent_coding()) return 1;
  if (test_buf_adopt_deleter()) return 1;
  std::printf("h2_test: %d checks passed\n", tests_run);
  return 0;
}
