// Pooled pinned-staging pipeline for device<->pageable-host transfers,
// shared between the HIP kernel extension (_native) and the C++ gRPC
// transport (_transport).
//
// Host-only HIP runtime API (no kernels), so it compiles under plain g++
// with -D__HIP_PLATFORM_AMD__ and links against amdhip64.
//
// Design (round-2 rework of the round-1 single-mutex StagingPool — see
// VERDICT.md "What's weak" #6):
//  * a POOL of independent staging contexts, each with its own HIP stream,
//    two pinned chunk buffers and two events. Concurrent callers (gRPC
//    worker threads, shm server threads) each lease their own context, so
//    D2H/H2D transfers from different requests overlap instead of
//    serializing behind one global mutex;
//  * the chunk consumer/producer is a callback, not a fixed memcpy: the
//    transport streams chunks straight into HTTP/2 DATA frames while the
//    NEXT chunk is still DMAing — the "device->pinned-host hipMemcpyAsync
//    overlaps protobuf encode and the gRPC send on a side HIP stream"
//    overlap of the north star (BASELINE.json), with the socket write as
//    the consumer;
//  * producer ordering is an EVENT recorded on the producing stream that
//    the staging stream waits on device-side (hipStreamWaitEvent), not a
//    blanket host-side hipStreamSynchronize.
#pragma once

#include <hip/hip_runtime_api.h>

#include <chrono>
#include <condition_variable>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <mutex>
#include <stdexcept>
#include <string>
#include <utility>
#include <vector>

namespace mi355x_staging {

#define MI355X_STAGING_CHECK(expr)                                       \
  do {                                                                   \
    hipError_t _e = (expr);                                              \
    if (_e != hipSuccess)                                                \
      throw std::runtime_error(std::string("HIP error in staging: ") +   \
                               hipGetErrorString(_e) + " at " #expr);    \
  } while (0)

// 4 MiB chunks x 2 buffers by default: deep enough to hide DMA behind
// the consumer (socket write / wire-buffer memcpy), small enough that
// the first byte reaches the consumer ~80 us after the transfer starts.
// Tunable via MI355X_STAGING_CHUNK (bytes, clamped to [1 MiB, 32 MiB],
// read once at first use) for A/B sweeps.
inline size_t chunk_size() {
  static const size_t v = [] {
    const char* e = std::getenv("MI355X_STAGING_CHUNK");
    long long n = e ? std::atoll(e) : 0;
    if (n < (1 << 20) || n > (32 << 20)) n = 4 << 20;
    return size_t(n);
  }();
  return v;
}
constexpr int kMaxCtx = 8;

struct Ctx {
  hipStream_t stream = nullptr;
  void* buf[2] = {nullptr, nullptr};
  hipEvent_t evt[2] = {};
  hipEvent_t producer_evt = {};

  void init() {
    MI355X_STAGING_CHECK(
        hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
    for (int i = 0; i < 2; ++i) {
      MI355X_STAGING_CHECK(hipHostMalloc(&buf[i], chunk_size()));
      MI355X_STAGING_CHECK(
          hipEventCreateWithFlags(&evt[i], hipEventDisableTiming));
    }
    MI355X_STAGING_CHECK(
        hipEventCreateWithFlags(&producer_evt, hipEventDisableTiming));
  }

  // Order this context's staging stream after all work currently queued
  // on `producer` (device-side wait; the host does not block).
  void wait_producer(hipStream_t producer) {
    MI355X_STAGING_CHECK(hipEventRecord(producer_evt, producer));
    MI355X_STAGING_CHECK(hipStreamWaitEvent(stream, producer_evt, 0));
  }

  // device -> consumer(pinned_chunk, len), pipelined 2-deep: while the
  // consumer drains chunk i, chunk i+1 is already DMAing into the other
  // pinned buffer.
  //
  // Overlap proof: with MI355X_STAGING_TRACE=<path> set, each d2h call
  // appends one JSON line with per-chunk {wait_ms, consume_ms} — wait_ms
  // is the host-visible time hipEventSynchronize blocked before the
  // chunk's DMA was done. After chunk 0, wait_ms ~ 0 means the DMA of
  // chunk i+1 fully overlapped the consumer (socket write / memcpy) of
  // chunk i; a non-overlapped pipeline would wait the full chunk DMA
  // time (~70 us at 4 MiB) on every chunk.
  template <typename Consume>
  void d2h(const void* src_dev, size_t nbytes, Consume&& consume) {
    const size_t kChunk = chunk_size();
    size_t nchunks = (nbytes + kChunk - 1) / kChunk;
    const char* trace_path = std::getenv("MI355X_STAGING_TRACE");
    std::vector<double> wait_ms, consume_ms;
    size_t issued = 0;
    for (size_t c = 0; c < (nchunks < 2 ? nchunks : 2); ++c) {
      size_t off = c * kChunk;
      size_t len = nbytes - off < kChunk ? nbytes - off : kChunk;
      MI355X_STAGING_CHECK(hipMemcpyAsync(
          buf[c & 1], static_cast<const char*>(src_dev) + off, len,
          hipMemcpyDeviceToHost, stream));
      MI355X_STAGING_CHECK(hipEventRecord(evt[c & 1], stream));
      ++issued;
    }
    for (size_t c = 0; c < nchunks; ++c) {
      size_t off = c * kChunk;
      size_t len = nbytes - off < kChunk ? nbytes - off : kChunk;
      std::chrono::steady_clock::time_point t0, t1, t2;
      if (trace_path) t0 = std::chrono::steady_clock::now();
      MI355X_STAGING_CHECK(hipEventSynchronize(evt[c & 1]));
      if (trace_path) t1 = std::chrono::steady_clock::now();
      consume(static_cast<const void*>(buf[c & 1]), len);
      if (trace_path) {
        t2 = std::chrono::steady_clock::now();
        wait_ms.push_back(
            std::chrono::duration<double, std::milli>(t1 - t0).count());
        consume_ms.push_back(
            std::chrono::duration<double, std::milli>(t2 - t1).count());
      }
      if (issued < nchunks) {
        size_t noff = issued * kChunk;
        size_t nlen = nbytes - noff < kChunk ? nbytes - noff : kChunk;
        MI355X_STAGING_CHECK(hipMemcpyAsync(
            buf[issued & 1], static_cast<const char*>(src_dev) + noff, nlen,
            hipMemcpyDeviceToHost, stream));
        MI355X_STAGING_CHECK(hipEventRecord(evt[issued & 1], stream));
        ++issued;
      }
    }
    if (trace_path && !wait_ms.empty()) {
      // one JSON line per d2h call; appends race-free enough via O_APPEND
      // line writes (trace is a diagnostic, not a hot-path feature)
      FILE* f = std::fopen(trace_path, "a");
      if (f) {
        std::fprintf(f, "{\"nbytes\": %zu, \"chunks\": %zu, \"wait_ms\": [",
                     nbytes, nchunks);
        for (size_t i = 0; i < wait_ms.size(); ++i)
          std::fprintf(f, "%s%.4f", i ? "," : "", wait_ms[i]);
        std::fprintf(f, "], \"consume_ms\": [");
        for (size_t i = 0; i < consume_ms.size(); ++i)
          std::fprintf(f, "%s%.4f", i ? "," : "", consume_ms[i]);
        std::fprintf(f, "]}\n");
        std::fclose(f);
      }
    }
  }

  // produce(pinned_chunk, len) fills each chunk; its H2D DMA overlaps the
  // production of the next chunk. Returns after the last DMA completes.
  template <typename Produce>
  void h2d(void* dst_dev, size_t nbytes, Produce&& produce) {
    const size_t kChunk = chunk_size();
    size_t nchunks = (nbytes + kChunk - 1) / kChunk;
    for (size_t c = 0; c < nchunks; ++c) {
      size_t off = c * kChunk;
      size_t len = nbytes - off < kChunk ? nbytes - off : kChunk;
      if (c >= 2) MI355X_STAGING_CHECK(hipEventSynchronize(evt[c & 1]));
      produce(buf[c & 1], len);
      MI355X_STAGING_CHECK(hipMemcpyAsync(
          static_cast<char*>(dst_dev) + off, buf[c & 1], len,
          hipMemcpyHostToDevice, stream));
      MI355X_STAGING_CHECK(hipEventRecord(evt[c & 1], stream));
    }
    MI355X_STAGING_CHECK(hipStreamSynchronize(stream));
  }
};

class Pool {
 public:
  static Pool& instance() {
    static Pool pool;
    return pool;
  }

  Ctx* acquire() {
    std::unique_lock<std::mutex> lk(mu_);
    cv_.wait(lk, [this] { return !free_.empty() || created_ < kMaxCtx; });
    if (!free_.empty()) {
      Ctx* c = free_.back();
      free_.pop_back();
      return c;
    }
    ++created_;
    lk.unlock();
    auto* c = new Ctx();
    try {
      c->init();
    } catch (...) {
      delete c;
      std::lock_guard<std::mutex> lk2(mu_);
      --created_;
      cv_.notify_one();
      throw;
    }
    return c;
  }

  void release(Ctx* c) {
    {
      std::lock_guard<std::mutex> lk(mu_);
      free_.push_back(c);
    }
    cv_.notify_one();
  }

 private:
  std::mutex mu_;
  std::condition_variable cv_;
  std::vector<Ctx*> free_;
  int created_ = 0;
};

// RAII lease over a pooled context.
struct Lease {
  Ctx* ctx;
  Lease() : ctx(Pool::instance().acquire()) {}
  ~Lease() { Pool::instance().release(ctx); }
  Lease(const Lease&) = delete;
  Lease& operator=(const Lease&) = delete;
};

}  // namespace mi355x_staging
