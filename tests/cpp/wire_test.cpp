// Standalone unit test for the hand-rolled wire codec (ops/csrc/wire.h) —
// no torch, no python. Built and run under ASAN/UBSAN by
// tools/run_sanitizers.sh (the SURVEY §5 sanitizer-CI requirement): the
// codec is the component that parses untrusted network bytes, so it is the
// one that must be memory-safe under malformed input.
#include <cassert>
#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

#include "../../min_tfs_client_amd/ops/csrc/wire.h"

using namespace tfswire;

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

int test_varint() {
  uint8_t buf[16];
  for (uint64_t v : {0ull, 1ull, 127ull, 128ull, 300ull, 1ull << 32,
                     ~0ull}) {
    uint8_t* end = write_varint(buf, v);
    CHECK(end - buf == varint_size(v));
    Cursor c{buf, end};
    CHECK(c.read_varint() == v);
    CHECK(c.done());
  }
  return 0;
}

int test_roundtrip() {
  std::vector<std::string> names = {"x", "attention_mask"};
  std::vector<TensorMeta> metas = {
      {1, {2, 3}, 24},          // fp32 2x3
      {3, {128, 512}, 128 * 512 * 4},
  };
  auto plan = plan_predict_message(true, "bert", 7, "serving_default",
                                   names, metas);
  std::vector<uint8_t> buf(plan.total_size);
  write_predict_message(buf.data(), plan, true, "bert", 7,
                        "serving_default", names, metas);
  // fill payloads with patterns
  for (size_t i = 0; i < plan.spans.size(); ++i) {
    std::memset(buf.data() + plan.spans[i].offset, int('A' + i),
                plan.spans[i].nbytes);
  }
  auto parsed = parse_predict_message(buf.data(), buf.size(), true);
  CHECK(parsed.model_spec.name == "bert");
  CHECK(parsed.model_spec.version == 7);
  CHECK(parsed.model_spec.signature_name == "serving_default");
  CHECK(parsed.tensors.size() == 2);
  CHECK(parsed.tensors[0].name == "x");
  CHECK(parsed.tensors[0].dtype == 1);
  CHECK(parsed.tensors[0].shape == (std::vector<int64_t>{2, 3}));
  CHECK(parsed.tensors[0].content_bytes == 24);
  CHECK(parsed.tensors[0].content[0] == 'A');
  CHECK(parsed.tensors[1].content_bytes == 128 * 512 * 4);
  CHECK(parsed.tensors[1].content[5] == 'B');
  return 0;
}

int test_response_field_numbers() {
  // response: map=1, spec=2 (predict.proto:35-40)
  std::vector<std::string> names = {"y"};
  std::vector<TensorMeta> metas = {{1, {1}, 4}};
  auto plan = plan_predict_message(false, "m", -1, "", names, metas);
  std::vector<uint8_t> buf(plan.total_size);
  write_predict_message(buf.data(), plan, false, "m", -1, "", names,
                        metas);
  CHECK((buf[0] >> 3) == 2);  // first field written: model_spec at 2
  auto parsed = parse_predict_message(buf.data(), buf.size(), false);
  CHECK(parsed.tensors.size() == 1 && parsed.tensors[0].name == "y");
  return 0;
}

int test_truncated_inputs_do_not_overrun() {
  std::vector<std::string> names = {"x"};
  std::vector<TensorMeta> metas = {{1, {1024}, 4096}};
  auto plan = plan_predict_message(true, "m", 1, "", names, metas);
  std::vector<uint8_t> buf(plan.total_size);
  write_predict_message(buf.data(), plan, true, "m", 1, "", names, metas);
  // every truncation point must throw, never read past the end
  for (size_t cut = 0; cut < 64; ++cut) {
    std::vector<uint8_t> t(buf.begin(), buf.begin() + cut);
    try {
      parse_predict_message(t.data(), t.size(), true);
    } catch (const std::exception&) {
      // expected for most cuts
    }
  }
  // malformed varints
  std::vector<uint8_t> evil(16, 0xFF);
  try {
    parse_predict_message(evil.data(), evil.size(), true);
  } catch (const std::exception&) {
  }
  return 0;
}

int test_typed_field_parse() {
  // hand-build: outputs entry with packed float_val [1.5, -2]
  // TensorProto: dtype=1 (08 01) + float_val packed (2a 08 <8 bytes>)
  uint8_t tp[] = {0x08, 0x01, 0x2a, 0x08,
                  0x00, 0x00, 0xc0, 0x3f, 0x00, 0x00, 0x00, 0xc0};
  std::vector<uint8_t> msg;
  msg.push_back(0x0a);  // outputs entry, field 1
  msg.push_back(uint8_t(3 + 2 + sizeof(tp)));
  msg.push_back(0x0a); msg.push_back(1); msg.push_back('z');  // key
  msg.push_back(0x12); msg.push_back(uint8_t(sizeof(tp)));
  msg.insert(msg.end(), tp, tp + sizeof(tp));
  auto parsed = parse_predict_message(msg.data(), msg.size(), false);
  CHECK(parsed.tensors.size() == 1);
  CHECK(parsed.tensors[0].floats.size() == 2);
  CHECK(parsed.tensors[0].floats[0] == 1.5f);
  CHECK(parsed.tensors[0].floats[1] == -2.0f);
  return 0;
}

int test_version_label_parse() {
  // ModelSpec{name="m", version_label="prod"(field 4)}
  std::vector<uint8_t> msg = {0x0a, 0x0b, 0x0a, 0x01, 'm',
                              0x22, 0x04, 'p', 'r', 'o', 'd', 0x00};
  msg[1] = 0x09;  // spec length: name(3) + label(6)
  msg.resize(11);
  auto parsed = parse_predict_message(msg.data(), msg.size(), true);
  CHECK(parsed.model_spec.name == "m");
  CHECK(parsed.model_spec.version_label == "prod");
  return 0;
}

int test_unknown_fields_skipped() {
  // field 99 varint, then a valid model_spec
  std::vector<uint8_t> msg = {0xd8, 0x06, 0x07,       // field 99 = 7
                              0x0a, 0x03, 0x0a, 0x01, 'm'};
  auto parsed = parse_predict_message(msg.data(), msg.size(), true);
  CHECK(parsed.model_spec.name == "m");
  return 0;
}

int main() {
  int rc = 0;
  rc |= test_varint();
  rc |= test_roundtrip();
  rc |= test_response_field_numbers();
  rc |= test_truncated_inputs_do_not_overrun();
  rc |= test_typed_field_parse();
  rc |= test_version_label_parse();
  rc |= test_unknown_fields_skipped();
  if (rc == 0) std::printf("wire_test: %d checks passed\n", tests_run);
  return rc;
}
