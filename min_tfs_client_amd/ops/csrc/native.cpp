// pybind module: fast PredictRequest/PredictResponse codec + HIP pack ops.
//
// The serialize path writes protobuf wire bytes directly from tensor memory
// (wire.h skeleton + DMA into the wire buffer) — no python-protobuf, no
// intermediate TensorProto objects. This replaces the reference's per-element
// python loop (reference tensors.py:17-25) and its server-side C++
// AsProtoField/AsProtoTensorContent pair (reference tensor.cc:948-965) with
// one MI355X-native path.

#include <torch/extension.h>

#include <algorithm>
#include <atomic>
#include <chrono>
#include <cstring>
#include <string>
#include <thread>
#include <vector>

#include "wire.h"

namespace py = pybind11;

// implemented in pack_kernels.hip
namespace mi355x {
at::Tensor cast_op(const at::Tensor& in, at::ScalarType out_dtype);
at::Tensor nchw_to_nhwc(const at::Tensor& in, at::ScalarType out_dtype);
at::Tensor nhwc_to_nchw(const at::Tensor& in, at::ScalarType out_dtype);
at::Tensor quantize_q8(const at::Tensor& in, double scale, double zp);
at::Tensor dequantize_q8(const at::Tensor& in, double scale, double zp);
void copy_device_to_host_ptr(const at::Tensor& src, void* dst, size_t n,
                             int mode);
void copy_host_ptr_to_device(const void* src, at::Tensor& dst, size_t n,
                             int mode);
bool hip_available();
}  // namespace mi355x

namespace {

// torch scalar type -> tensorflow.DataType enum (types.proto:12-68)
int torch_to_tf_dtype(at::ScalarType t) {
  switch (t) {
    case at::kFloat: return 1;
    case at::kDouble: return 2;
    case at::kInt: return 3;
    case at::kByte: return 4;
    case at::kShort: return 5;
    case at::kChar: return 6;
    case at::kComplexFloat: return 8;
    case at::kLong: return 9;
    case at::kBool: return 10;
    case at::kBFloat16: return 14;
    case at::kUInt16: return 17;
    case at::kComplexDouble: return 18;
    case at::kHalf: return 19;
    case at::kUInt32: return 22;
    case at::kUInt64: return 23;
    default:
      throw std::runtime_error("unsupported torch dtype for TensorProto");
  }
}

at::ScalarType tf_to_torch_dtype(int dt) {
  switch (dt) {
    case 1: return at::kFloat;
    case 2: return at::kDouble;
    case 3: return at::kInt;
    case 4: return at::kByte;
    case 5: return at::kShort;
    case 6: return at::kChar;
    case 8: return at::kComplexFloat;
    case 9: return at::kLong;
    case 10: return at::kBool;
    case 14: return at::kBFloat16;
    case 17: return at::kUInt16;
    case 18: return at::kComplexDouble;
    case 19: return at::kHalf;
    case 22: return at::kUInt32;
    case 23: return at::kUInt64;
    default:
      throw std::runtime_error("unsupported TensorProto dtype " +
                               std::to_string(dt));
  }
}

// ---------------------------------------------------------------------------
// serialize (client request / server response)
// ---------------------------------------------------------------------------

py::bytes serialize_predict(bool is_request, const std::string& model_name,
                            int64_t version, const std::string& signature,
                            const std::vector<std::string>& names,
                            const std::vector<at::Tensor>& tensors,
                            int copy_mode) {
  TORCH_CHECK(names.size() == tensors.size(), "names/tensors mismatch");
  std::vector<tfswire::TensorMeta> metas(tensors.size());
  std::vector<at::Tensor> contig(tensors.size());
  for (size_t i = 0; i < tensors.size(); ++i) {
    contig[i] = tensors[i].contiguous();
    metas[i].dtype = torch_to_tf_dtype(contig[i].scalar_type());
    auto sizes = contig[i].sizes();
    metas[i].shape.assign(sizes.begin(), sizes.end());
    metas[i].content_bytes =
        uint64_t(contig[i].numel()) * contig[i].element_size();
    // protobuf bytes fields cap at 2GB (TF enforces the same limit)
    TORCH_CHECK(metas[i].content_bytes < (uint64_t(1) << 31),
                "tensor '", names[i], "' exceeds the 2GB tensor_content "
                "limit of the protobuf wire format");
  }
  auto plan = tfswire::plan_predict_message(is_request, model_name, version,
                                            signature, names, metas);
  // allocate the final bytes object and write into it in place
  PyObject* obj = PyBytes_FromStringAndSize(nullptr,
                                            Py_ssize_t(plan.total_size));
  if (!obj) throw std::bad_alloc();
  auto* buf = reinterpret_cast<uint8_t*>(PyBytes_AS_STRING(obj));
  tfswire::write_predict_message(buf, plan, is_request, model_name, version,
                                 signature, names, metas);
  {
    py::gil_scoped_release release;
    for (size_t i = 0; i < contig.size(); ++i) {
      const auto& span = plan.spans[i];
      if (span.nbytes == 0) continue;
      if (contig[i].is_cuda()) {
        mi355x::copy_device_to_host_ptr(contig[i], buf + span.offset,
                                        span.nbytes, copy_mode);
      } else {
        std::memcpy(buf + span.offset, contig[i].const_data_ptr(),
                    span.nbytes);
      }
    }
  }
  return py::reinterpret_steal<py::bytes>(obj);
}

// Streaming serialize: writes only the wire SKELETON (everything except
// large tensor_content payloads) and returns the payload locations as
// regions, so the transport can stream them — pinned-staging DMA chunks
// straight into DATA frames for device tensors (copy/send overlap), iovec
// directly from tensor memory for host tensors (true zero-copy send, the
// analogue of the reference's two-slice encode,
// grpc_tensor_coding.cc:140-248).
//
// Returns (skeleton_bytes, regions, keepalive):
//   regions  = [(offset, nbytes, ptr, is_device), ...] ascending offsets;
//   keepalive = the contiguous tensors backing `ptr` — the caller must
//   hold it until the send completes.
// Host spans smaller than kStreamMinSpan are copied into the skeleton
// (fewer iovec regions beats the copy at that size).
constexpr size_t kStreamMinSpan = 64 * 1024;

py::tuple serialize_predict_streaming(bool is_request,
                                      const std::string& model_name,
                                      int64_t version,
                                      const std::string& signature,
                                      const std::vector<std::string>& names,
                                      const std::vector<at::Tensor>& tensors) {
  TORCH_CHECK(names.size() == tensors.size(), "names/tensors mismatch");
  std::vector<tfswire::TensorMeta> metas(tensors.size());
  std::vector<at::Tensor> contig(tensors.size());
  for (size_t i = 0; i < tensors.size(); ++i) {
    contig[i] = tensors[i].contiguous();
    metas[i].dtype = torch_to_tf_dtype(contig[i].scalar_type());
    auto sizes = contig[i].sizes();
    metas[i].shape.assign(sizes.begin(), sizes.end());
    metas[i].content_bytes =
        uint64_t(contig[i].numel()) * contig[i].element_size();
    TORCH_CHECK(metas[i].content_bytes < (uint64_t(1) << 31),
                "tensor '", names[i], "' exceeds the 2GB tensor_content "
                "limit of the protobuf wire format");
  }
  auto plan = tfswire::plan_predict_message(is_request, model_name, version,
                                            signature, names, metas);
  PyObject* obj = PyBytes_FromStringAndSize(nullptr,
                                            Py_ssize_t(plan.total_size));
  if (!obj) throw std::bad_alloc();
  auto* buf = reinterpret_cast<uint8_t*>(PyBytes_AS_STRING(obj));
  tfswire::write_predict_message(buf, plan, is_request, model_name, version,
                                 signature, names, metas);
  py::list regions;
  py::list keepalive;
  // plan.spans arrive in tensor order; region list must ascend by offset
  std::vector<size_t> order(contig.size());
  for (size_t i = 0; i < order.size(); ++i) order[i] = i;
  std::sort(order.begin(), order.end(), [&](size_t a, size_t b) {
    return plan.spans[a].offset < plan.spans[b].offset;
  });
  for (size_t idx : order) {
    const auto& span = plan.spans[idx];
    if (span.nbytes == 0) continue;
    const at::Tensor& t = contig[idx];
    if (t.is_cuda()) {
      regions.append(py::make_tuple(
          uint64_t(span.offset), uint64_t(span.nbytes),
          uint64_t(reinterpret_cast<uintptr_t>(t.const_data_ptr())), true));
      keepalive.append(t);
    } else if (span.nbytes >= kStreamMinSpan) {
      regions.append(py::make_tuple(
          uint64_t(span.offset), uint64_t(span.nbytes),
          uint64_t(reinterpret_cast<uintptr_t>(t.const_data_ptr())), false));
      keepalive.append(t);
    } else {
      std::memcpy(buf + span.offset, t.const_data_ptr(), span.nbytes);
    }
  }
  return py::make_tuple(py::reinterpret_steal<py::bytes>(obj), regions,
                        keepalive);
}

// ---------------------------------------------------------------------------
// parse (client response / server request)
// ---------------------------------------------------------------------------

// Builds a torch tensor from a parsed TensorProto span. `device`: "" or
// "cpu" => CPU tensor; "cuda"/"cuda:N" => device tensor via the staging
// pipeline.
at::Tensor tensor_from_parsed(const tfswire::ParsedTensor& t,
                              const std::string& device, int copy_mode) {
  at::ScalarType st = tf_to_torch_dtype(t.dtype);
  std::vector<int64_t> shape = t.shape;
  int64_t numel = 1;
  for (auto d : shape) numel *= d;
  const bool to_cuda = device.rfind("cuda", 0) == 0;
  auto cpu_opts = at::TensorOptions().dtype(st).device(at::kCPU);

  at::Tensor cpu;
  if (t.content != nullptr) {
    int64_t elem = int64_t(c10::elementSize(st));
    TORCH_CHECK(int64_t(t.content_bytes) >= numel * elem,
                "tensor_content too short");
    if (to_cuda) {
      auto dev = at::empty(shape, at::TensorOptions().dtype(st).device(
                                      at::Device(device)));
      py::gil_scoped_release release;
      mi355x::copy_host_ptr_to_device(t.content, dev,
                                      size_t(numel * elem), copy_mode);
      return dev;
    }
    cpu = at::empty(shape, cpu_opts);
    py::gil_scoped_release release;
    std::memcpy(cpu.mutable_data_ptr(), t.content, size_t(numel * elem));
    return cpu;
  }
  // typed-field fallback (incl. TF repeat-last-fill, tensor.cc:487-527)
  cpu = at::empty(shape, cpu_opts);
  auto fill = [&](auto* dst, const auto& src) {
    using D = std::remove_pointer_t<decltype(dst)>;
    int64_t n = int64_t(src.size());
    for (int64_t i = 0; i < numel; ++i) {
      int64_t j = i < n ? i : (n > 0 ? n - 1 : 0);
      dst[i] = n > 0 ? D(src[size_t(j)]) : D(0);
    }
  };
  switch (st) {
    case at::kFloat: fill(cpu.data_ptr<float>(), t.floats); break;
    case at::kDouble: fill(cpu.data_ptr<double>(), t.doubles); break;
    case at::kInt: fill(cpu.data_ptr<int32_t>(), t.ints); break;
    case at::kLong: fill(cpu.data_ptr<int64_t>(), t.ints); break;
    case at::kShort: fill(cpu.data_ptr<int16_t>(), t.ints); break;
    case at::kChar: fill(cpu.data_ptr<int8_t>(), t.ints); break;
    case at::kByte: fill(cpu.data_ptr<uint8_t>(), t.ints); break;
    case at::kBool: fill(cpu.data_ptr<bool>(), t.ints); break;
    case at::kComplexFloat: case at::kComplexDouble: {
      // scomplex_val/dcomplex_val: interleaved re/im pairs
      // (tensor.proto:59-61); repeat-last-fill repeats the last PAIR
      const bool single = st == at::kComplexFloat;
      int64_t pairs = single ? int64_t(t.floats.size()) / 2
                             : int64_t(t.doubles.size()) / 2;
      if (single) {
        auto* dst = reinterpret_cast<float*>(cpu.data_ptr());
        for (int64_t i = 0; i < numel; ++i) {
          int64_t j = i < pairs ? i : (pairs > 0 ? pairs - 1 : 0);
          dst[2 * i] = pairs > 0 ? t.floats[size_t(2 * j)] : 0.f;
          dst[2 * i + 1] = pairs > 0 ? t.floats[size_t(2 * j + 1)] : 0.f;
        }
      } else {
        auto* dst = reinterpret_cast<double*>(cpu.data_ptr());
        for (int64_t i = 0; i < numel; ++i) {
          int64_t j = i < pairs ? i : (pairs > 0 ? pairs - 1 : 0);
          dst[2 * i] = pairs > 0 ? t.doubles[size_t(2 * j)] : 0.0;
          dst[2 * i + 1] = pairs > 0 ? t.doubles[size_t(2 * j + 1)] : 0.0;
        }
      }
      break;
    }
    case at::kHalf: case at::kBFloat16: {
      // half_val holds raw uint16 bit-patterns (tensor.cc:446-464)
      auto* dst = reinterpret_cast<uint16_t*>(cpu.data_ptr());
      int64_t n = int64_t(t.ints.size());
      for (int64_t i = 0; i < numel; ++i) {
        int64_t j = i < n ? i : (n > 0 ? n - 1 : 0);
        dst[i] = n > 0 ? uint16_t(t.ints[size_t(j)]) : 0;
      }
      break;
    }
    default:
      TORCH_CHECK(false, "typed-field decode unsupported for this dtype");
  }
  if (to_cuda) return cpu.to(at::Device(device));
  return cpu;
}

py::tuple parse_predict(py::buffer data, bool is_request,
                        const std::string& device, int copy_mode) {
  py::buffer_info info = data.request();
  auto parsed = tfswire::parse_predict_message(
      static_cast<const uint8_t*>(info.ptr), size_t(info.size), is_request);
  py::dict out;
  for (auto& t : parsed.tensors) {
    if (t.dtype == 7) {  // DT_STRING
      py::list vals;
      for (auto& s : t.strings) vals.append(py::bytes(s));
      out[py::str(t.name)] = vals;
    } else {
      out[py::str(t.name)] = tensor_from_parsed(t, device, copy_mode);
    }
  }
  py::dict spec;
  spec["name"] = parsed.model_spec.name;
  spec["version"] = parsed.model_spec.version;
  spec["signature_name"] = parsed.model_spec.signature_name;
  spec["version_label"] = parsed.model_spec.version_label;
  py::list filt;
  for (auto& f : parsed.output_filter) filt.append(py::str(f));
  return py::make_tuple(spec, out, filt);
}

// Server echo fast path: parse the request bytes and build the identity
// response entirely in C++ (payload memcpy host->host), mirroring the
// reference fixture model's semantics (*_input -> *_output aliases).
py::bytes echo_predict(py::buffer data) {
  py::buffer_info info = data.request();
  auto parsed = tfswire::parse_predict_message(
      static_cast<const uint8_t*>(info.ptr), size_t(info.size), true);
  std::vector<std::string> names;
  std::vector<tfswire::TensorMeta> metas;
  std::vector<const uint8_t*> payloads;
  for (auto& t : parsed.tensors) {
    TORCH_CHECK(t.content != nullptr,
                "echo_predict requires tensor_content inputs");
    std::string name = t.name;
    const std::string suffix = "_input";
    if (name.size() > suffix.size() &&
        name.compare(name.size() - suffix.size(), suffix.size(), suffix)
            == 0) {
      name = name.substr(0, name.size() - suffix.size()) + "_output";
    }
    names.push_back(std::move(name));
    metas.push_back({t.dtype, t.shape, t.content_bytes});
    payloads.push_back(t.content);
  }
  auto plan = tfswire::plan_predict_message(
      false, parsed.model_spec.name, parsed.model_spec.version,
      parsed.model_spec.signature_name.empty()
          ? "serving_default" : parsed.model_spec.signature_name,
      names, metas);
  PyObject* obj = PyBytes_FromStringAndSize(nullptr,
                                            Py_ssize_t(plan.total_size));
  if (!obj) throw std::bad_alloc();
  auto* buf = reinterpret_cast<uint8_t*>(PyBytes_AS_STRING(obj));
  tfswire::write_predict_message(
      buf, plan, false, parsed.model_spec.name, parsed.model_spec.version,
      parsed.model_spec.signature_name.empty()
          ? "serving_default" : parsed.model_spec.signature_name,
      names, metas);
  {
    py::gil_scoped_release release;
    for (size_t i = 0; i < payloads.size(); ++i) {
      std::memcpy(buf + plan.spans[i].offset, payloads[i],
                  plan.spans[i].nbytes);
    }
  }
  return py::reinterpret_steal<py::bytes>(obj);
}

}  // namespace

// echo directly between two buffers (shm identity fast path, the
// counterpart of echo_predict for the gRPC raw handler): parses the
// request from src, writes the response into dst, memcpying payloads
// once. Returns response length.
uint64_t echo_predict_into(py::buffer src, uint64_t src_len,
                           py::buffer dst) {
  py::buffer_info si = src.request();
  py::buffer_info di = dst.request(true);
  TORCH_CHECK(uint64_t(si.size) >= src_len, "src too short");
  auto parsed = tfswire::parse_predict_message(
      static_cast<const uint8_t*>(si.ptr), size_t(src_len), true);
  std::vector<std::string> names;
  std::vector<tfswire::TensorMeta> metas;
  std::vector<const uint8_t*> payloads;
  for (auto& t : parsed.tensors) {
    TORCH_CHECK(t.content != nullptr,
                "echo_predict_into requires tensor_content inputs");
    std::string name = t.name;
    const std::string suffix = "_input";
    if (name.size() > suffix.size() &&
        name.compare(name.size() - suffix.size(), suffix.size(), suffix)
            == 0) {
      name = name.substr(0, name.size() - suffix.size()) + "_output";
    }
    names.push_back(std::move(name));
    metas.push_back({t.dtype, t.shape, t.content_bytes});
    payloads.push_back(t.content);
  }
  const std::string sig = parsed.model_spec.signature_name.empty()
      ? "serving_default" : parsed.model_spec.signature_name;
  auto plan = tfswire::plan_predict_message(
      false, parsed.model_spec.name, parsed.model_spec.version, sig,
      names, metas);
  TORCH_CHECK(plan.total_size <= uint64_t(di.size),
              "dst slot too small for echo response");
  auto* buf = static_cast<uint8_t*>(di.ptr);
  tfswire::write_predict_message(buf, plan, false, parsed.model_spec.name,
                                 parsed.model_spec.version, sig, names,
                                 metas);
  {
    py::gil_scoped_release release;
    for (size_t i = 0; i < payloads.size(); ++i) {
      std::memcpy(buf + plan.spans[i].offset, payloads[i],
                  plan.spans[i].nbytes);
    }
  }
  return plan.total_size;
}

// serialize into a caller-provided buffer (shared-memory transport):
// returns bytes written; throws if capacity is too small.
uint64_t serialize_predict_into(py::buffer dst, bool is_request,
                                const std::string& model_name,
                                int64_t version,
                                const std::string& signature,
                                const std::vector<std::string>& names,
                                const std::vector<at::Tensor>& tensors,
                                int copy_mode) {
  TORCH_CHECK(names.size() == tensors.size(), "names/tensors mismatch");
  py::buffer_info info = dst.request(true);
  std::vector<tfswire::TensorMeta> metas(tensors.size());
  std::vector<at::Tensor> contig(tensors.size());
  for (size_t i = 0; i < tensors.size(); ++i) {
    contig[i] = tensors[i].contiguous();
    metas[i].dtype = torch_to_tf_dtype(contig[i].scalar_type());
    auto sizes = contig[i].sizes();
    metas[i].shape.assign(sizes.begin(), sizes.end());
    metas[i].content_bytes =
        uint64_t(contig[i].numel()) * contig[i].element_size();
    TORCH_CHECK(metas[i].content_bytes < (uint64_t(1) << 31),
                "tensor exceeds the 2GB tensor_content limit");
  }
  auto plan = tfswire::plan_predict_message(is_request, model_name, version,
                                            signature, names, metas);
  TORCH_CHECK(plan.total_size <= uint64_t(info.size),
              "shm slot too small: need ", plan.total_size, " bytes, have ",
              info.size);
  auto* buf = static_cast<uint8_t*>(info.ptr);
  tfswire::write_predict_message(buf, plan, is_request, model_name, version,
                                 signature, names, metas);
  {
    py::gil_scoped_release release;
    for (size_t i = 0; i < contig.size(); ++i) {
      const auto& span = plan.spans[i];
      if (span.nbytes == 0) continue;
      if (contig[i].is_cuda()) {
        mi355x::copy_device_to_host_ptr(contig[i], buf + span.offset,
                                        span.nbytes, copy_mode);
      } else {
        std::memcpy(buf + span.offset, contig[i].const_data_ptr(),
                    span.nbytes);
      }
    }
  }
  return plan.total_size;
}

// GIL-released wait on a shared u32 (x86 cross-process visibility):
// returns the observed value, or 0xFFFFFFFF on timeout.
uint32_t shm_wait_value(py::buffer buf, uint64_t offset, uint32_t target,
                        double timeout_s) {
  py::buffer_info info = buf.request();
  // C++-conformant cross-process visibility: atomic acquire load (the
  // round-1 volatile+fence worked on x86-64 but was UB-adjacent)
  auto* p = reinterpret_cast<const std::atomic<uint32_t>*>(
      static_cast<uint8_t*>(info.ptr) + offset);
  static_assert(sizeof(std::atomic<uint32_t>) == sizeof(uint32_t),
                "atomic<u32> must be layout-compatible with u32");
  py::gil_scoped_release release;
  const auto deadline = std::chrono::steady_clock::now() +
      std::chrono::duration<double>(timeout_s);
  int spins = 0;
  while (true) {
    uint32_t v = p->load(std::memory_order_acquire);
    if (v == target) {
      return v;
    }
    if (std::chrono::steady_clock::now() > deadline) return 0xFFFFFFFFu;
    if (++spins < 1024) {
#if defined(__x86_64__)
      __builtin_ia32_pause();
#endif
    } else {
      std::this_thread::sleep_for(std::chrono::microseconds(
          spins < 4096 ? 5 : 50));
    }
  }
}

void shm_store_value(py::buffer buf, uint64_t offset, uint32_t value) {
  py::buffer_info info = buf.request(true);
  auto* p = reinterpret_cast<std::atomic<uint32_t>*>(
      static_cast<uint8_t*>(info.ptr) + offset);
  p->store(value, std::memory_order_release);
}

// Zero-copy span parse: returns per-tensor (dtype, shape, offset, length)
// into the SOURCE buffer, no tensor construction — the python side wraps
// views with torch.frombuffer, borrowing the response bytes' memory.
py::tuple parse_predict_spans(py::buffer data, bool is_request) {
  py::buffer_info info = data.request();
  auto parsed = tfswire::parse_predict_message(
      static_cast<const uint8_t*>(info.ptr), size_t(info.size), is_request);
  py::list spans;
  for (auto& t : parsed.tensors) {
    if (t.content == nullptr) {
      spans.append(py::none());  // typed-field tensor: caller falls back
      continue;
    }
    py::dict d;
    d["name"] = t.name;
    d["dtype"] = t.dtype;
    d["shape"] = t.shape;
    d["offset"] = uint64_t(t.content -
                           static_cast<const uint8_t*>(info.ptr));
    d["nbytes"] = t.content_bytes;
    spans.append(d);
  }
  py::dict spec;
  spec["name"] = parsed.model_spec.name;
  spec["version"] = parsed.model_spec.version;
  spec["signature_name"] = parsed.model_spec.signature_name;
  spec["version_label"] = parsed.model_spec.version_label;
  py::list filt;
  for (auto& f : parsed.output_filter) filt.append(py::str(f));
  return py::make_tuple(spec, spans, filt);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X-native TF-Serving codec + CDNA4 pack kernels";
  m.def("serialize_predict_request",
        [](const std::string& model_name, int64_t version,
           const std::string& signature,
           const std::vector<std::string>& names,
           const std::vector<at::Tensor>& tensors, int copy_mode) {
          return serialize_predict(true, model_name, version, signature,
                                   names, tensors, copy_mode);
        },
        py::arg("model_name"), py::arg("version"), py::arg("signature"),
        py::arg("names"), py::arg("tensors"), py::arg("copy_mode") = 0);
  m.def("serialize_predict_response",
        [](const std::string& model_name, int64_t version,
           const std::string& signature,
           const std::vector<std::string>& names,
           const std::vector<at::Tensor>& tensors, int copy_mode) {
          return serialize_predict(false, model_name, version, signature,
                                   names, tensors, copy_mode);
        },
        py::arg("model_name"), py::arg("version"), py::arg("signature"),
        py::arg("names"), py::arg("tensors"), py::arg("copy_mode") = 0);
  m.def("serialize_predict_streaming", &serialize_predict_streaming,
        py::arg("is_request"), py::arg("model_name"), py::arg("version"),
        py::arg("signature"), py::arg("names"), py::arg("tensors"),
        "Skeleton + payload-region streaming serialize: returns (bytes, "
        "regions[(offset,nbytes,ptr,is_device)], keepalive) for the "
        "transport's overlapped/zero-copy send path.");
  m.def("parse_predict_request",
        [](py::buffer b, const std::string& device, int copy_mode) {
          return parse_predict(b, true, device, copy_mode);
        },
        py::arg("data"), py::arg("device") = "cpu", py::arg("copy_mode") = 0);
  m.def("parse_predict_response",
        [](py::buffer b, const std::string& device, int copy_mode) {
          return parse_predict(b, false, device, copy_mode);
        },
        py::arg("data"), py::arg("device") = "cpu", py::arg("copy_mode") = 0);
  m.def("echo_predict", &echo_predict, py::arg("data"));
  m.def("parse_predict_spans", &parse_predict_spans, py::arg("data"),
        py::arg("is_request") = false);
  m.def("echo_predict_into", &echo_predict_into, py::arg("src"),
        py::arg("src_len"), py::arg("dst"));
  m.def("serialize_predict_into", &serialize_predict_into,
        py::arg("dst"), py::arg("is_request"), py::arg("model_name"),
        py::arg("version"), py::arg("signature"), py::arg("names"),
        py::arg("tensors"), py::arg("copy_mode") = 1);
  m.def("shm_wait_value", &shm_wait_value, py::arg("buf"),
        py::arg("offset"), py::arg("target"), py::arg("timeout_s"));
  m.def("shm_store_value", &shm_store_value, py::arg("buf"),
        py::arg("offset"), py::arg("value"));
  m.def("tensor_content_bytes",
        [](const at::Tensor& t, int copy_mode) {
          auto c = t.contiguous();
          size_t n = size_t(c.numel()) * c.element_size();
          PyObject* obj = PyBytes_FromStringAndSize(nullptr, Py_ssize_t(n));
          if (!obj) throw std::bad_alloc();
          auto* buf = PyBytes_AS_STRING(obj);
          {
            py::gil_scoped_release release;
            if (c.is_cuda()) {
              mi355x::copy_device_to_host_ptr(c, buf, n, copy_mode);
            } else {
              std::memcpy(buf, c.const_data_ptr(), n);
            }
          }
          return py::reinterpret_steal<py::bytes>(obj);
        },
        py::arg("tensor"), py::arg("copy_mode") = 0);
  // HIP ops
  m.def("cast", &mi355x::cast_op, py::arg("input"), py::arg("out_dtype"));
  m.def("nchw_to_nhwc", &mi355x::nchw_to_nhwc, py::arg("input"),
        py::arg("out_dtype"));
  m.def("nhwc_to_nchw", &mi355x::nhwc_to_nchw, py::arg("input"),
        py::arg("out_dtype"));
  m.def("quantize_q8", &mi355x::quantize_q8, py::arg("input"),
        py::arg("scale"), py::arg("zero_point") = 0.0);
  m.def("dequantize_q8", &mi355x::dequantize_q8, py::arg("input"),
        py::arg("scale"), py::arg("zero_point") = 0.0);
  m.def("hip_available", &mi355x::hip_available);
}
