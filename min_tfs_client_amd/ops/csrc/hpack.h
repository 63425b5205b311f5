// HPACK (RFC 7541) header compression for the native gRPC transport.
//
// The decoder is complete (static + dynamic table, huffman strings, table
// size updates) because standard gRPC peers (grpcio's chttp2) emit
// incrementally-indexed huffman-coded literals. The encoder is deliberately
// minimal: indexed static entries where one exists, literal-without-indexing
// (no huffman) otherwise — always legal, never requires peer state.
//
// This replaces the reference's dependency on grpc's own chttp2 HPACK
// (the reference client rides python-grpcio) with a
// from-scratch implementation specialized for unary Predict traffic.
#pragma once

#include <cstdint>
#include <cstring>
#include <deque>
#include <stdexcept>
#include <string>
#include <utility>
#include <vector>

#include "hpack_huffman_table.h"

namespace h2 {

using Header = std::pair<std::string, std::string>;

// ---------------------------------------------------------------------------
// static table (RFC 7541 Appendix A) — 1-based indices 1..61
// ---------------------------------------------------------------------------
struct StaticEntry { const char* name; const char* value; };
inline constexpr StaticEntry kStaticTable[62] = {
    {"", ""},  // index 0 unused
    {":authority", ""},
    {":method", "GET"},
    {":method", "POST"},
    {":path", "/"},
    {":path", "/index.html"},
    {":scheme", "http"},
    {":scheme", "https"},
    {":status", "200"},
    {":status", "204"},
    {":status", "206"},
    {":status", "304"},
    {":status", "400"},
    {":status", "404"},
    {":status", "500"},
    {"accept-charset", ""},
    {"accept-encoding", "gzip, deflate"},
    {"accept-language", ""},
    {"accept-ranges", ""},
    {"accept", ""},
    {"access-control-allow-origin", ""},
    {"age", ""},
    {"allow", ""},
    {"authorization", ""},
    {"cache-control", ""},
    {"content-disposition", ""},
    {"content-encoding", ""},
    {"content-language", ""},
    {"content-length", ""},
    {"content-location", ""},
    {"content-range", ""},
    {"content-type", ""},
    {"cookie", ""},
    {"date", ""},
    {"etag", ""},
    {"expect", ""},
    {"expires", ""},
    {"from", ""},
    {"host", ""},
    {"if-match", ""},
    {"if-modified-since", ""},
    {"if-none-match", ""},
    {"if-range", ""},
    {"if-unmodified-since", ""},
    {"last-modified", ""},
    {"link", ""},
    {"location", ""},
    {"max-forwards", ""},
    {"proxy-authenticate", ""},
    {"proxy-authorization", ""},
    {"range", ""},
    {"referer", ""},
    {"refresh", ""},
    {"retry-after", ""},
    {"server", ""},
    {"set-cookie", ""},
    {"strict-transport-security", ""},
    {"transfer-encoding", ""},
    {"user-agent", ""},
    {"vary", ""},
    {"via", ""},
    {"www-authenticate", ""},
};

// ---------------------------------------------------------------------------
// huffman decoding — binary trie built once from the RFC code table
// ---------------------------------------------------------------------------
class HuffmanDecoder {
 public:
  static const HuffmanDecoder& instance() {
    static HuffmanDecoder d;
    return d;
  }

  // appends decoded bytes to `out`; throws on invalid padding/codes
  void decode(const uint8_t* p, size_t n, std::string* out) const {
    int node = 0;
    int bits_since_symbol = 0;
    for (size_t i = 0; i < n; ++i) {
      uint8_t byte = p[i];
      for (int b = 7; b >= 0; --b) {
        int bit = (byte >> b) & 1;
        node = nodes_[size_t(node)].next[bit];
        if (node < 0) throw std::runtime_error("hpack: bad huffman code");
        ++bits_since_symbol;
        int sym = nodes_[size_t(node)].symbol;
        if (sym >= 0) {
          if (sym == 256) throw std::runtime_error("hpack: EOS in string");
          out->push_back(char(sym));
          node = 0;
          bits_since_symbol = 0;
        }
      }
    }
    // remaining bits must be a prefix of EOS (all ones), < 8 bits
    if (bits_since_symbol >= 8)
      throw std::runtime_error("hpack: huffman padding too long");
    // walking 1-bits from the last partial node must stay on the EOS path;
    // since padding is the MSBs of EOS (all ones), any incomplete node we
    // stopped at is valid iff we got here by 1-bits only. Verify by checking
    // the node is on the all-ones path from root.
    int check = 0;
    for (int i = 0; i < bits_since_symbol; ++i) {
      check = nodes_[size_t(check)].next[1];
      if (check < 0) throw std::runtime_error("hpack: bad huffman padding");
    }
    if (check != node) throw std::runtime_error("hpack: bad huffman padding");
  }

 private:
  struct Node {
    int next[2] = {-1, -1};
    int symbol = -1;
  };
  std::vector<Node> nodes_;

  HuffmanDecoder() {
    nodes_.reserve(8192);  // total code bits = 4688; node count < that
    nodes_.emplace_back();  // root
    for (int sym = 0; sym <= 256; ++sym) {
      uint32_t code = kHuffCodes[sym];
      int len = kHuffLens[sym];
      int node = 0;
      for (int b = len - 1; b >= 0; --b) {
        int bit = int((code >> b) & 1);
        // NOTE: take a value, not a reference — emplace_back may
        // reallocate nodes_ and a held reference would dangle
        int nxt = nodes_[size_t(node)].next[bit];
        if (nxt < 0) {
          nxt = int(nodes_.size());
          nodes_.emplace_back();
          nodes_[size_t(node)].next[bit] = nxt;
        }
        node = nxt;
      }
      nodes_[size_t(node)].symbol = sym;
    }
  }
};

// huffman encoding (used for large header values like long :path strings —
// optional; callers may always use raw literals instead)
inline void huffman_encode(const std::string& in, std::string* out) {
  uint64_t acc = 0;
  int nbits = 0;
  for (unsigned char ch : in) {
    acc = (acc << kHuffLens[ch]) | kHuffCodes[ch];
    nbits += kHuffLens[ch];
    while (nbits >= 8) {
      nbits -= 8;
      out->push_back(char(uint8_t(acc >> nbits)));
    }
  }
  if (nbits > 0) {  // pad with EOS prefix (all ones)
    out->push_back(char(uint8_t((acc << (8 - nbits)) | ((1u << (8 - nbits)) - 1))));
  }
}

// ---------------------------------------------------------------------------
// integer primitives (RFC 7541 §5.1)
// ---------------------------------------------------------------------------
inline void encode_int(uint64_t v, int prefix_bits, uint8_t first_byte_flags,
                       std::string* out) {
  uint64_t max_prefix = (uint64_t(1) << prefix_bits) - 1;
  if (v < max_prefix) {
    out->push_back(char(first_byte_flags | uint8_t(v)));
    return;
  }
  out->push_back(char(first_byte_flags | uint8_t(max_prefix)));
  v -= max_prefix;
  while (v >= 128) {
    out->push_back(char(uint8_t(v & 0x7f) | 0x80));
    v >>= 7;
  }
  out->push_back(char(uint8_t(v)));
}

struct ByteCursor {
  const uint8_t* p;
  const uint8_t* end;
  bool done() const { return p >= end; }
  uint8_t peek() const {
    if (p >= end) throw std::runtime_error("hpack: truncated");
    return *p;
  }
  uint8_t next() {
    if (p >= end) throw std::runtime_error("hpack: truncated");
    return *p++;
  }
  uint64_t decode_int(int prefix_bits) {
    uint64_t max_prefix = (uint64_t(1) << prefix_bits) - 1;
    uint64_t v = next() & max_prefix;
    if (v < max_prefix) return v;
    int shift = 0;
    while (true) {
      uint8_t b = next();
      v += uint64_t(b & 0x7f) << shift;
      shift += 7;
      if (!(b & 0x80)) return v;
      if (shift > 56) throw std::runtime_error("hpack: integer overflow");
    }
  }
  std::string decode_string() {
    bool huff = (peek() & 0x80) != 0;
    uint64_t len = decode_int(7);
    if (uint64_t(end - p) < len) throw std::runtime_error("hpack: truncated");
    std::string s;
    if (huff) {
      s.reserve(size_t(len) * 2);
      HuffmanDecoder::instance().decode(p, size_t(len), &s);
    } else {
      s.assign(reinterpret_cast<const char*>(p), size_t(len));
    }
    p += len;
    return s;
  }
};

// ---------------------------------------------------------------------------
// decoder with dynamic table
// ---------------------------------------------------------------------------
class HpackDecoder {
 public:
  explicit HpackDecoder(size_t max_table_size = 4096)
      : max_size_(max_table_size), settings_max_size_(max_table_size) {}

  // SETTINGS_HEADER_TABLE_SIZE from our SETTINGS governs the peer encoder's
  // allowed maximum; the peer may shrink below it with a table-size update.
  void set_settings_max_size(size_t n) {
    settings_max_size_ = n;
    if (max_size_ > n) {
      max_size_ = n;
      evict();
    }
  }

  std::vector<Header> decode(const uint8_t* data, size_t n) {
    std::vector<Header> out;
    ByteCursor c{data, data + n};
    while (!c.done()) {
      uint8_t b = c.peek();
      if (b & 0x80) {  // indexed header field
        uint64_t idx = c.decode_int(7);
        out.push_back(lookup(idx));
      } else if (b & 0x40) {  // literal with incremental indexing
        uint64_t idx = c.decode_int(6);
        Header h = read_literal(idx, &c);
        insert(h);
        out.push_back(std::move(h));
      } else if (b & 0x20) {  // dynamic table size update
        uint64_t sz = c.decode_int(5);
        if (sz > settings_max_size_)
          throw std::runtime_error("hpack: table size update above limit");
        max_size_ = size_t(sz);
        evict();
      } else {  // literal without indexing (0x00) / never indexed (0x10)
        uint64_t idx = c.decode_int(4);
        out.push_back(read_literal(idx, &c));
      }
    }
    return out;
  }

 private:
  std::deque<Header> dyn_;       // most recent at front
  size_t dyn_size_ = 0;          // RFC size: name+value+32 per entry
  size_t max_size_;
  size_t settings_max_size_;

  static size_t entry_size(const Header& h) {
    return h.first.size() + h.second.size() + 32;
  }

  Header lookup(uint64_t idx) const {
    if (idx == 0) throw std::runtime_error("hpack: index 0");
    if (idx <= 61)
      return {kStaticTable[idx].name, kStaticTable[idx].value};
    size_t di = size_t(idx - 62);
    if (di >= dyn_.size()) throw std::runtime_error("hpack: bad index");
    return dyn_[di];
  }

  Header read_literal(uint64_t name_idx, ByteCursor* c) {
    Header h;
    if (name_idx == 0) {
      h.first = c->decode_string();
    } else {
      h.first = lookup(name_idx).first;
    }
    h.second = c->decode_string();
    return h;
  }

  void insert(const Header& h) {
    size_t sz = entry_size(h);
    if (sz > max_size_) {  // entry larger than table: table is emptied
      dyn_.clear();
      dyn_size_ = 0;
      return;
    }
    dyn_.push_front(h);
    dyn_size_ += sz;
    evict();
  }

  void evict() {
    while (dyn_size_ > max_size_ && !dyn_.empty()) {
      dyn_size_ -= entry_size(dyn_.back());
      dyn_.pop_back();
    }
  }
};

// ---------------------------------------------------------------------------
// encoder — stateless literals only (plus static-table indexed fields)
// ---------------------------------------------------------------------------
class HpackEncoder {
 public:
  // well-known fully-indexed fields
  void add_indexed(std::string* out, int static_index) {
    encode_int(uint64_t(static_index), 7, 0x80, out);
  }

  // literal without indexing; name_idx > 0 references the static table
  void add_literal(std::string* out, int static_name_index,
                   const std::string& value, bool huffman = false) {
    encode_int(uint64_t(static_name_index), 4, 0x00, out);
    write_string(out, value, huffman);
  }

  void add_literal(std::string* out, const std::string& name,
                   const std::string& value, bool huffman = false) {
    out->push_back(0x00);
    write_string(out, name, huffman);
    write_string(out, value, huffman);
  }

 private:
  static void write_string(std::string* out, const std::string& s,
                           bool huffman) {
    if (huffman) {
      std::string enc;
      enc.reserve(s.size());
      huffman_encode(s, &enc);
      if (enc.size() < s.size()) {
        encode_int(enc.size(), 7, 0x80, out);
        out->append(enc);
        return;
      }
    }
    encode_int(s.size(), 7, 0x00, out);
    out->append(s);
  }
};

}  // namespace h2
