// core.hpp — engine foundation: error handling, context, device memory,
// SyncedMemory and Blob.
//
// MI355X-native re-design of the reference memory substrate
// (include/caffe/syncedmem.hpp:39 head-state machine, blob.hpp data+diff
// pair, util/gpu_memory.hpp caching allocator).  Differences by design:
//  - fp32 only in round 1; the Type enum keeps the reference's surface
//    (type.hpp:13-47) so mixed precision can land later.
//  - every allocation is padded to PAD_ELEMS floats (the reference pads to
//    even counts, tensor.cpp:43-51, so flat NCCL calls never overrun; we pad
//    wider to keep 32B-aligned vector loads legal on every blob edge).
//  - one process drives one GPU (bench/train) — the reference's
//    thread-per-GPU P2PSync collapses to per-process streams + RCCL.
#pragma once

#include <algorithm>
#include <atomic>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <map>
#include <memory>
#include <mutex>
#include <random>
#include <sstream>
#include <stdexcept>
#include <string>
#include <vector>

#include <hip/hip_runtime.h>

namespace camd {

// ---------------------------------------------------------------- logging
// error-context breadcrumb: Net sets this to "layer 'conv1' (Convolution)
// Reshape" etc. before dispatching into layer code, so a CHECK failure
// deep in Blob/kernel plumbing names the layer it fired under (round-1
// review: a bare "CHECK failed: (d) >= (0)" was undiagnosable)
const char*& error_context();

struct FatalStream {
  std::ostringstream ss;
  const char* file;
  int line;
  FatalStream(const char* f, int l) : file(f), line(l) {}
  [[noreturn]] ~FatalStream() noexcept(false) {
    std::string msg = ss.str();
    if (error_context()) {
      msg += " [in ";
      msg += error_context();
      msg += "]";
    }
    fprintf(stderr, "[caffe_amd FATAL %s:%d] %s\n", file, line, msg.c_str());
    throw std::runtime_error(msg);
  }
  template <typename T>
  FatalStream& operator<<(const T& v) {
    ss << v;
    return *this;
  }
};

#define CAMD_FATAL ::camd::FatalStream(__FILE__, __LINE__)
#define CHECK_(c) \
  if (!(c)) CAMD_FATAL << "CHECK failed: " #c " "
#define CHECK_EQ_(a, b) CHECK_((a) == (b)) << "(" << (a) << " vs " << (b) << ") "
#define CHECK_NE_(a, b) CHECK_((a) != (b))
#define CHECK_GT_(a, b) CHECK_((a) > (b))
#define CHECK_GE_(a, b) CHECK_((a) >= (b))
#define CHECK_LT_(a, b) CHECK_((a) < (b))
#define CHECK_LE_(a, b) CHECK_((a) <= (b))

#define HIP_CHECK(x)                                                      \
  do {                                                                    \
    hipError_t e_ = (x);                                                  \
    if (e_ != hipSuccess)                                                 \
      CAMD_FATAL << "HIP error: " << hipGetErrorString(e_) << " at " #x;  \
  } while (0)

// ---------------------------------------------------------------- context
enum class Phase { TRAIN = 0, TEST = 1 };
enum class Mode { CPU = 0, GPU = 1 };

constexpr int PAD_ELEMS = 16;  // pad blob counts to 16 floats (64 B)
inline size_t padded(size_t count) {
  return (count + PAD_ELEMS - 1) / PAD_ELEMS * PAD_ELEMS;
}

// shared counter-based RNG mixer (identical on host and device so CPU and
// GPU synthetic data / dropout masks match bit-for-bit given one key)
inline uint64_t h_splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}
inline float h_u01(uint64_t h) {
  return (float)((h >> 11) * (1.0 / 9007199254740992.0));
}

// Simple caching device allocator (replaces the reference's CUB
// CachingDeviceAllocator, util/gpu_memory.cpp:144): frees go to a size-keyed
// free list, reused exactly; device memory is plentiful (288 GB HBM3E).
class DeviceAllocator {
 public:
  void* alloc(size_t bytes);
  void release(void* p, size_t bytes);
  void free_all();
  ~DeviceAllocator() { free_all(); }

 private:
  std::multimap<size_t, void*> free_;
  std::mutex mu_;
};

struct PerfClass {
  std::atomic<long> launches{0};
  std::atomic<double> flops{0};
  std::atomic<double> bytes{0};
  std::atomic<double> ns{0};  // filled only when timing enabled
  void add(double fl, double by) {
    launches.fetch_add(1, std::memory_order_relaxed);
    // atomic<double>::fetch_add needs C++20; emulate
    for (double cur = flops.load(); !flops.compare_exchange_weak(cur, cur + fl);) {}
    for (double cur = bytes.load(); !bytes.compare_exchange_weak(cur, cur + by);) {}
  }
  void add_ns(double v) {
    for (double cur = ns.load(); !ns.compare_exchange_weak(cur, cur + v);) {}
  }
};

// Engine: per-process global context (the reference's Caffe singleton,
// common.hpp:334-345 per-thread streams; here one process == one device).
class Engine {
 public:
  static Engine& get();

  Mode mode = Mode::CPU;
  int device = 0;
  uint64_t seed = 1371;  // `caffe time` pins 1371 (tools/caffe.cpp:365)
  int rank = 0;          // solver rank; fillers use seed+rank
  int world = 1;

  // GEMM compute dtype: false = fp32 MFMA (exact, the parity/default
  // path), true = bf16 MFMA with fp32 accumulation (mixed precision —
  // the reference's Ftype/Btype fp16 analog).  Storage
  // stays fp32 either way; only the conv/IP contractions change.
  bool gemm_bf16 = false;

  // synthetic-data configuration (no LMDB datasets in this environment)
  bool synthetic = true;
  int syn_classes = 1000;
  // current training iteration (set by Solver::Step) — synthetic data and
  // dropout masks key off this so snapshot/restore resumes the exact
  // stream (the reference's LMDB cursor position analog)
  uint64_t data_iter = 0;

  bool gpu_inited = false;
  hipStream_t stream = nullptr;       // compute stream
  hipStream_t comm_stream = nullptr;  // collectives + updates (side stream)

  void set_mode_gpu(int dev);
  void sync() {
    if (mode == Mode::GPU) {
      HIP_CHECK(hipStreamSynchronize(stream));
      HIP_CHECK(hipStreamSynchronize(comm_stream));
    }
  }

  DeviceAllocator dalloc;
  std::mt19937_64 cpu_rng{1371};

  // perf registry: kernel-class name -> counters
  std::map<std::string, PerfClass>& perf() { return perf_; }
  PerfClass& perf(const std::string& k) {
    std::lock_guard<std::mutex> g(perf_mu_);
    return perf_[k];
  }
  bool perf_timing = false;  // when true, kernels get event pairs

  struct EvPair {
    hipEvent_t a, b;
    PerfClass* pc;
  };
  std::vector<EvPair> pending_events;
  void drain_events();  // call after a sync

 private:
  Engine() = default;
  std::map<std::string, PerfClass> perf_;
  std::mutex perf_mu_;
};

// RAII perf timer for a kernel launch region on a stream.  The string
// constructor resolves the class each call; hot launchers should pass a
// cached PerfClass* (PERF_CLASS macro) — the registry lookup + string
// construction otherwise runs ~2000x per training step on the host path.
struct PerfScope {
  PerfClass* pc = nullptr;
  hipStream_t s = nullptr;
  hipEvent_t a = nullptr, b = nullptr;
  PerfScope(PerfClass* cls, hipStream_t stream, double flops, double bytes) {
    Engine& E = Engine::get();
    pc = cls;
    pc->add(flops, bytes);
    if (E.perf_timing && E.mode == Mode::GPU) {
      s = stream;
      HIP_CHECK(hipEventCreate(&a));
      HIP_CHECK(hipEventCreate(&b));
      HIP_CHECK(hipEventRecord(a, s));
    }
  }
  PerfScope(const std::string& name, hipStream_t stream, double flops,
            double bytes)
      : PerfScope(&Engine::get().perf(name), stream, flops, bytes) {}
  ~PerfScope() {
    if (a) {
      HIP_CHECK(hipEventRecord(b, s));
      Engine::get().pending_events.push_back({a, b, pc});
    }
  }
};
// per-call-site cached perf class (static local: resolved once)
#define PERF_CLASS(name)                                        \
  ([]() -> ::camd::PerfClass* {                                 \
    static ::camd::PerfClass* pc_ = &::camd::Engine::get().perf(name); \
    return pc_;                                                 \
  }())

// ---------------------------------------------------------- SyncedMemory
// Lazy host/device mirror with the reference's head-state machine
// (syncedmem.hpp:39: UNINITIALIZED / HEAD_AT_CPU / HEAD_AT_GPU / SYNCED).
class SyncedMemory {
 public:
  enum Head { UNINIT, AT_CPU, AT_GPU, SYNCED };

  explicit SyncedMemory(size_t bytes) : bytes_(bytes) {}
  ~SyncedMemory();

  const void* cpu_data();
  const void* gpu_data();
  void* mutable_cpu_data();
  void* mutable_gpu_data();
  // Raw device pointer without state transition (for views)
  void* gpu_ptr_raw() { return gpu_ptr_; }
  size_t size() const { return bytes_; }
  Head head() const { return head_; }

  // Make this memory a non-owning view into an external device arena
  // (the reference's Blob::set_gpu_diff into the learnable-diff space,
  // blob.hpp:487 / net.cpp:1369).
  void set_gpu_view(void* p);

 private:
  void to_cpu();
  void to_gpu();
  size_t bytes_;
  void* cpu_ptr_ = nullptr;
  void* gpu_ptr_ = nullptr;
  bool own_gpu_ = false;
  Head head_ = UNINIT;
};

// ------------------------------------------------------------------ Blob
// data + diff, runtime-shaped (reference blob.hpp; fp32 only this round).
class Blob {
 public:
  Blob() = default;
  explicit Blob(const std::vector<int>& shape) { Reshape(shape); }

  void Reshape(const std::vector<int>& shape);
  void ReshapeLike(const Blob& o) { Reshape(o.shape_); }

  const std::vector<int>& shape() const { return shape_; }
  int shape(int i) const {
    if (i < 0) i += (int)shape_.size();
    return shape_[i];
  }
  int num_axes() const { return (int)shape_.size(); }
  long count() const { return count_; }
  long count(int start) const {
    long c = 1;
    for (int i = start; i < (int)shape_.size(); ++i) c *= shape_[i];
    return c;
  }
  long count(int start, int end) const {
    long c = 1;
    for (int i = start; i < end; ++i) c *= shape_[i];
    return c;
  }
  int num() const { return shape_.empty() ? 1 : shape_[0]; }
  int channels() const { return num_axes() > 1 ? shape_[1] : 1; }
  int height() const { return num_axes() > 2 ? shape_[2] : 1; }
  int width() const { return num_axes() > 3 ? shape_[3] : 1; }

  const float* cpu_data() { return (const float*)data_->cpu_data(); }
  const float* cpu_diff() { return (const float*)diff_->cpu_data(); }
  float* mutable_cpu_data() { return (float*)data_->mutable_cpu_data(); }
  float* mutable_cpu_diff() { return (float*)diff_->mutable_cpu_data(); }
  const float* gpu_data() { return (const float*)data_->gpu_data(); }
  const float* gpu_diff() { return (const float*)diff_->gpu_data(); }
  float* mutable_gpu_data() { return (float*)data_->mutable_gpu_data(); }
  float* mutable_gpu_diff() { return (float*)diff_->mutable_gpu_data(); }

  SyncedMemory& data_mem() { return *data_; }
  SyncedMemory& diff_mem() { return *diff_; }
  // Share another blob's data (SplitLayer / in-place layers / test-net
  // weight sharing — reference Blob::ShareData, blob.hpp)
  void ShareData(Blob& o) { data_ = o.data_; }
  void ShareDiff(Blob& o) { diff_ = o.diff_; }

  void set_data_const(float v);
  void set_diff_const(float v);

  std::string shape_string() const {
    std::ostringstream s;
    for (int d : shape_) s << d << " ";
    s << "(" << count_ << ")";
    return s.str();
  }

 private:
  std::vector<int> shape_;
  long count_ = 0;
  std::shared_ptr<SyncedMemory> data_, diff_;
};

}  // namespace camd
