// Hand-rolled protobuf wire codec for the TF-Serving Predict hot path.
//
// Emits/parses PredictRequest / PredictResponse bytes directly (no protobuf
// library): the serializer computes the full message skeleton up front and
// returns the byte offsets where each tensor's `tensor_content` payload must
// land, so device->host DMA can write *straight into the wire buffer* — the
// MI355X-native evolution of TF's two-slice zero-copy encode
// (reference grpc_tensor_coding.cc:93-246, EncodeSkeleton + payload slice).
//
// Wire facts (field numbers) from:
//   /root/reference/protobuf_srcs/tensorflow_serving/apis/predict.proto:12-40
//   /root/reference/protobuf_srcs/tensorflow_serving/apis/model.proto:9-33
//   /root/reference/protobuf_srcs/tensorflow/core/framework/tensor.proto:14-94
//   /root/reference/protobuf_srcs/tensorflow/core/framework/tensor_shape.proto:13-46
#pragma once

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace tfswire {

// ---------------------------------------------------------------------------
// varint / tag primitives
// ---------------------------------------------------------------------------

inline int varint_size(uint64_t v) {
  int n = 1;
  while (v >= 0x80) { v >>= 7; ++n; }
  return n;
}

inline uint8_t* write_varint(uint8_t* p, uint64_t v) {
  while (v >= 0x80) { *p++ = static_cast<uint8_t>(v) | 0x80; v >>= 7; }
  *p++ = static_cast<uint8_t>(v);
  return p;
}

// tag = (field_number << 3) | wire_type
inline int tag_size(int field) { return varint_size(uint64_t(field) << 3); }
inline uint8_t* write_tag(uint8_t* p, int field, int wt) {
  return write_varint(p, (uint64_t(field) << 3) | uint64_t(wt));
}

constexpr int WT_VARINT = 0;
constexpr int WT_I64 = 1;
constexpr int WT_LEN = 2;
constexpr int WT_I32 = 5;

inline int len_delim_size(int field, uint64_t payload) {
  return tag_size(field) + varint_size(payload) + int(payload);
}

// ---------------------------------------------------------------------------
// sizes of the nested messages we emit
// ---------------------------------------------------------------------------

struct TensorMeta {
  int dtype;                       // tensorflow.DataType enum value
  std::vector<int64_t> shape;
  uint64_t content_bytes;          // length of tensor_content payload
};

// TensorShapeProto{ repeated Dim{int64 size=1} dim=2 }
inline uint64_t shape_proto_size(const std::vector<int64_t>& shape) {
  uint64_t total = 0;
  for (int64_t d : shape) {
    uint64_t dim_payload = 1 + varint_size(uint64_t(d));   // field1 varint
    total += 1 + varint_size(dim_payload) + dim_payload;   // field2 len-delim
  }
  return total;
}

inline uint8_t* write_shape_proto(uint8_t* p, const std::vector<int64_t>& s) {
  for (int64_t d : s) {
    uint64_t dim_payload = 1 + varint_size(uint64_t(d));
    p = write_tag(p, 2, WT_LEN);             // TensorShapeProto.dim
    p = write_varint(p, dim_payload);
    p = write_tag(p, 1, WT_VARINT);          // Dim.size
    p = write_varint(p, uint64_t(d));
  }
  return p;
}

// TensorProto{dtype=1, tensor_shape=2, tensor_content=4}
inline uint64_t tensor_proto_size(const TensorMeta& m) {
  uint64_t sz = 0;
  if (m.dtype != 0) sz += 1 + varint_size(uint64_t(m.dtype));
  uint64_t shp = shape_proto_size(m.shape);
  sz += 1 + varint_size(shp) + shp;          // tensor_shape (emit even if 0-d)
  sz += 1 + varint_size(m.content_bytes) + m.content_bytes;  // tensor_content
  return sz;
}

// Writes the TensorProto header; *content_offset receives the byte offset
// (relative to `base`) where the raw payload starts. The payload bytes are
// NOT written here — the caller DMA-copies them into place.
inline uint8_t* write_tensor_proto(uint8_t* p, const uint8_t* base,
                                   const TensorMeta& m,
                                   uint64_t* content_offset) {
  if (m.dtype != 0) {
    p = write_tag(p, 1, WT_VARINT);
    p = write_varint(p, uint64_t(m.dtype));
  }
  uint64_t shp = shape_proto_size(m.shape);
  p = write_tag(p, 2, WT_LEN);
  p = write_varint(p, shp);
  p = write_shape_proto(p, m.shape);
  p = write_tag(p, 4, WT_LEN);               // tensor_content
  p = write_varint(p, m.content_bytes);
  *content_offset = uint64_t(p - base);
  return p + m.content_bytes;                // skip payload region
}

// ModelSpec{name=1, Int64Value version=2, signature_name=3}
inline uint64_t model_spec_size(const std::string& name, int64_t version,
                                const std::string& signature) {
  uint64_t sz = 0;
  if (!name.empty()) sz += len_delim_size(1, name.size());
  if (version >= 0) {
    uint64_t iv = 1 + varint_size(uint64_t(version));  // Int64Value.value=1
    sz += 1 + varint_size(iv) + iv;
  }
  if (!signature.empty()) sz += len_delim_size(3, signature.size());
  return sz;
}

inline uint8_t* write_model_spec(uint8_t* p, const std::string& name,
                                 int64_t version,
                                 const std::string& signature) {
  if (!name.empty()) {
    p = write_tag(p, 1, WT_LEN);
    p = write_varint(p, name.size());
    std::memcpy(p, name.data(), name.size());
    p += name.size();
  }
  if (version >= 0) {
    uint64_t iv = 1 + varint_size(uint64_t(version));
    p = write_tag(p, 2, WT_LEN);
    p = write_varint(p, iv);
    p = write_tag(p, 1, WT_VARINT);
    p = write_varint(p, uint64_t(version));
  }
  if (!signature.empty()) {
    p = write_tag(p, 3, WT_LEN);
    p = write_varint(p, signature.size());
    std::memcpy(p, signature.data(), signature.size());
    p += signature.size();
  }
  return p;
}

// ---------------------------------------------------------------------------
// PredictRequest / PredictResponse skeleton serialization
// ---------------------------------------------------------------------------

struct SerializedSpan {
  uint64_t offset;       // where the tensor payload must be written
  uint64_t nbytes;
};

struct SkeletonPlan {
  uint64_t total_size = 0;
  std::vector<SerializedSpan> spans;   // one per tensor, input order
};

// map entry: key=1 (string), value=2 (TensorProto)
inline uint64_t map_entry_size(const std::string& key, uint64_t value_size) {
  return len_delim_size(1, key.size()) + 1 + varint_size(value_size) +
         value_size;
}

// Computes the full size of a PredictRequest (map_field=2 inputs) or
// PredictResponse (map_field=1 outputs; model_spec field differs too).
inline SkeletonPlan plan_predict_message(
    bool is_request, const std::string& model_name, int64_t version,
    const std::string& signature, const std::vector<std::string>& names,
    const std::vector<TensorMeta>& metas) {
  SkeletonPlan plan;
  const int spec_field = is_request ? 1 : 2;
  const int map_field = is_request ? 2 : 1;
  uint64_t total = 0;
  uint64_t spec = model_spec_size(model_name, version, signature);
  if (spec > 0 || !model_name.empty())
    total += tag_size(spec_field) + varint_size(spec) + spec;
  for (size_t i = 0; i < names.size(); ++i) {
    uint64_t tp = tensor_proto_size(metas[i]);
    uint64_t entry = map_entry_size(names[i], tp);
    total += tag_size(map_field) + varint_size(entry) + entry;
  }
  plan.total_size = total;
  plan.spans.resize(names.size());
  return plan;
}

// Writes the skeleton into buf (size from plan_predict_message) and fills
// plan.spans with the payload offsets.
inline void write_predict_message(
    uint8_t* buf, SkeletonPlan& plan, bool is_request,
    const std::string& model_name, int64_t version,
    const std::string& signature, const std::vector<std::string>& names,
    const std::vector<TensorMeta>& metas) {
  const int spec_field = is_request ? 1 : 2;
  const int map_field = is_request ? 2 : 1;
  uint8_t* p = buf;
  uint64_t spec = model_spec_size(model_name, version, signature);
  if (spec > 0 || !model_name.empty()) {
    p = write_tag(p, spec_field, WT_LEN);
    p = write_varint(p, spec);
    p = write_model_spec(p, model_name, version, signature);
  }
  for (size_t i = 0; i < names.size(); ++i) {
    uint64_t tp = tensor_proto_size(metas[i]);
    uint64_t entry = map_entry_size(names[i], tp);
    p = write_tag(p, map_field, WT_LEN);
    p = write_varint(p, entry);
    p = write_tag(p, 1, WT_LEN);                 // entry.key
    p = write_varint(p, names[i].size());
    std::memcpy(p, names[i].data(), names[i].size());
    p += names[i].size();
    p = write_tag(p, 2, WT_LEN);                 // entry.value
    p = write_varint(p, tp);
    uint64_t off = 0;
    p = write_tensor_proto(p, buf, metas[i], &off);
    plan.spans[i] = {off, metas[i].content_bytes};
  }
  if (uint64_t(p - buf) != plan.total_size)
    throw std::runtime_error("wire: skeleton size mismatch");
}

// ---------------------------------------------------------------------------
// Parser
// ---------------------------------------------------------------------------

struct Cursor {
  const uint8_t* p;
  const uint8_t* end;
  bool done() const { return p >= end; }
  uint64_t read_varint() {
    uint64_t v = 0; int shift = 0;
    while (true) {
      if (p >= end) throw std::runtime_error("wire: truncated varint");
      uint8_t b = *p++;
      v |= uint64_t(b & 0x7f) << shift;
      if (!(b & 0x80)) return v;
      shift += 7;
      if (shift > 63) throw std::runtime_error("wire: varint overflow");
    }
  }
  // returns field number, sets wire type
  int read_tag(int* wt) {
    uint64_t t = read_varint();
    *wt = int(t & 7);
    return int(t >> 3);
  }
  Cursor read_len_delim() {
    uint64_t n = read_varint();
    if (uint64_t(end - p) < n) throw std::runtime_error("wire: truncated");
    Cursor c{p, p + n};
    p += n;
    return c;
  }
  void skip(int wt) {
    switch (wt) {
      case WT_VARINT: read_varint(); break;
      case WT_I64:
        if (end - p < 8) throw std::runtime_error("wire: truncated");
        p += 8; break;
      case WT_LEN: read_len_delim(); break;
      case WT_I32:
        if (end - p < 4) throw std::runtime_error("wire: truncated");
        p += 4; break;
      default: throw std::runtime_error("wire: bad wire type");
    }
  }
};

struct ParsedTensor {
  std::string name;
  int dtype = 0;
  std::vector<int64_t> shape;
  // tensor_content span (into the source buffer); nullptr if typed fields
  const uint8_t* content = nullptr;
  uint64_t content_bytes = 0;
  // typed-field fallback: raw span of the packed/unpacked field bytes is not
  // exposed; instead decoded values land here (float/double/int collapsed to
  // int64/double domains is lossy, so keep per-kind vectors).
  std::vector<int64_t> ints;
  std::vector<float> floats;
  std::vector<double> doubles;
  std::vector<std::string> strings;
};

struct ParsedModelSpec {
  std::string name;
  int64_t version = -1;
  std::string signature_name;
  std::string version_label;   // ModelSpec field 4 (oneof with version)
};

inline void parse_shape(Cursor c, std::vector<int64_t>* shape) {
  while (!c.done()) {
    int wt; int f = c.read_tag(&wt);
    if (f == 2 && wt == WT_LEN) {       // dim
      Cursor d = c.read_len_delim();
      int64_t size = 0;
      while (!d.done()) {
        int dwt; int df = d.read_tag(&dwt);
        if (df == 1 && dwt == WT_VARINT) size = int64_t(d.read_varint());
        else d.skip(dwt);
      }
      shape->push_back(size);
    } else {
      c.skip(wt);
    }
  }
}

// Parses a TensorProto submessage (typed fields decoded for the val kinds
// the client protocol uses; tensor_content kept as a zero-copy span).
inline void parse_tensor_proto(Cursor c, ParsedTensor* t) {
  while (!c.done()) {
    int wt; int f = c.read_tag(&wt);
    switch (f) {
      case 1:  t->dtype = int(c.read_varint()); break;
      case 2:  parse_shape(c.read_len_delim(), &t->shape); break;
      case 4: {
        Cursor b = c.read_len_delim();
        t->content = b.p;
        t->content_bytes = uint64_t(b.end - b.p);
        break;
      }
      // float_val=5; scomplex_val=9 is interleaved re/im floats
      // (tensor.proto:59-61) — same wire shape, dtype disambiguates
      case 5: case 9: {
        if (wt == WT_LEN) {
          Cursor b = c.read_len_delim();
          // bound on >=4 so a malformed 4k+r-byte payload cannot read
          // past the submessage end; trailing bytes are an error
          while (b.end - b.p >= 4) {
            float v; std::memcpy(&v, b.p, 4); b.p += 4;
            t->floats.push_back(v);
          }
          if (b.p != b.end)
            throw std::runtime_error("wire: truncated packed float field");
        } else { uint32_t raw = 0;
          if (c.end - c.p < 4) throw std::runtime_error("wire: truncated");
          std::memcpy(&raw, c.p, 4); c.p += 4;
          float v; std::memcpy(&v, &raw, 4); t->floats.push_back(v); }
        break;
      }
      // double_val=6; dcomplex_val=12 is interleaved re/im doubles
      case 6: case 12: {
        if (wt == WT_LEN) {
          Cursor b = c.read_len_delim();
          while (b.end - b.p >= 8) {
            double v; std::memcpy(&v, b.p, 8); b.p += 8;
            t->doubles.push_back(v);
          }
          if (b.p != b.end)
            throw std::runtime_error("wire: truncated packed double field");
        } else { if (c.end - c.p < 8)
            throw std::runtime_error("wire: truncated");
          double v; std::memcpy(&v, c.p, 8); c.p += 8;
          t->doubles.push_back(v); }
        break;
      }
      case 7: case 10: case 11: case 13: case 16: case 17: {  // int kinds
        if (wt == WT_LEN) {
          Cursor b = c.read_len_delim();
          while (!b.done()) t->ints.push_back(int64_t(b.read_varint()));
        } else {
          t->ints.push_back(int64_t(c.read_varint()));
        }
        break;
      }
      case 8: {  // string_val
        Cursor b = c.read_len_delim();
        t->strings.emplace_back(reinterpret_cast<const char*>(b.p),
                                size_t(b.end - b.p));
        break;
      }
      default:
        c.skip(wt);
    }
  }
}

inline void parse_model_spec(Cursor c, ParsedModelSpec* m) {
  while (!c.done()) {
    int wt; int f = c.read_tag(&wt);
    if (f == 1 && wt == WT_LEN) {
      Cursor b = c.read_len_delim();
      m->name.assign(reinterpret_cast<const char*>(b.p),
                     size_t(b.end - b.p));
    } else if (f == 2 && wt == WT_LEN) {
      Cursor b = c.read_len_delim();
      while (!b.done()) {
        int iwt; int iff = b.read_tag(&iwt);
        if (iff == 1 && iwt == WT_VARINT) m->version = int64_t(b.read_varint());
        else b.skip(iwt);
      }
    } else if (f == 3 && wt == WT_LEN) {
      Cursor b = c.read_len_delim();
      m->signature_name.assign(reinterpret_cast<const char*>(b.p),
                               size_t(b.end - b.p));
    } else if (f == 4 && wt == WT_LEN) {
      Cursor b = c.read_len_delim();
      m->version_label.assign(reinterpret_cast<const char*>(b.p),
                              size_t(b.end - b.p));
    } else {
      c.skip(wt);
    }
  }
}

struct ParsedPredict {
  ParsedModelSpec model_spec;
  std::vector<ParsedTensor> tensors;
  std::vector<std::string> output_filter;
};

// is_request: map field = 2, model_spec = 1; response: map = 1, spec = 2.
inline ParsedPredict parse_predict_message(const uint8_t* data, size_t size,
                                           bool is_request) {
  ParsedPredict out;
  const int spec_field = is_request ? 1 : 2;
  const int map_field = is_request ? 2 : 1;
  Cursor c{data, data + size};
  while (!c.done()) {
    int wt; int f = c.read_tag(&wt);
    if (f == spec_field && wt == WT_LEN) {
      parse_model_spec(c.read_len_delim(), &out.model_spec);
    } else if (f == map_field && wt == WT_LEN) {
      Cursor e = c.read_len_delim();
      ParsedTensor t;
      while (!e.done()) {
        int ewt; int ef = e.read_tag(&ewt);
        if (ef == 1 && ewt == WT_LEN) {
          Cursor k = e.read_len_delim();
          t.name.assign(reinterpret_cast<const char*>(k.p),
                        size_t(k.end - k.p));
        } else if (ef == 2 && ewt == WT_LEN) {
          parse_tensor_proto(e.read_len_delim(), &t);
        } else {
          e.skip(ewt);
        }
      }
      out.tensors.push_back(std::move(t));
    } else if (is_request && f == 3 && wt == WT_LEN) {
      Cursor b = c.read_len_delim();
      out.output_filter.emplace_back(reinterpret_cast<const char*>(b.p),
                                     size_t(b.end - b.p));
    } else {
      c.skip(wt);
    }
  }
  return out;
}

}  // namespace tfswire
