#include "core.hpp"

namespace camd {

Engine& Engine::get() {
  static Engine e;
  return e;
}

const char*& error_context() {
  thread_local const char* ctx = nullptr;
  return ctx;
}

void Engine::set_mode_gpu(int dev) {
  mode = Mode::GPU;
  device = dev;
  if (!gpu_inited) {
    int n = 0;
    hipError_t err = hipGetDeviceCount(&n);
    if (err != hipSuccess || n == 0)
      CAMD_FATAL << "GPU mode requested but no HIP device is available "
                    "(the engine never falls back to CPU silently): "
                 << hipGetErrorString(err);
    HIP_CHECK(hipSetDevice(dev));
    HIP_CHECK(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
    HIP_CHECK(hipStreamCreateWithFlags(&comm_stream, hipStreamNonBlocking));
    gpu_inited = true;
  }
}

void Engine::drain_events() {
  for (auto& p : pending_events) {
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, p.a, p.b));
    p.pc->add_ns((double)ms * 1e6);
    HIP_CHECK(hipEventDestroy(p.a));
    HIP_CHECK(hipEventDestroy(p.b));
  }
  pending_events.clear();
}

void* DeviceAllocator::alloc(size_t bytes) {
  // +64B safety tail on EVERY device allocation: staged GEMM loads read
  // whole 16B lane chunks through per-image spad padding and trailing
  // partial chunks (values are dropped or masked, but the BYTES must be
  // mapped) — without slack the last row of a tensor can touch the next
  // page (observed as a GPU memory-access fault under the glds kernel)
  bytes += 64;
  {
    std::lock_guard<std::mutex> g(mu_);
    auto it = free_.find(bytes);
    if (it != free_.end()) {
      void* p = it->second;
      free_.erase(it);
      return p;
    }
  }
  void* p = nullptr;
  hipError_t e = hipMalloc(&p, bytes);
  if (e != hipSuccess) {
    free_all();  // return cached blocks and retry once
    HIP_CHECK(hipMalloc(&p, bytes));
  }
  return p;
}

void DeviceAllocator::release(void* p, size_t bytes) {
  bytes += 64;  // must mirror alloc()'s safety-tail key
  std::lock_guard<std::mutex> g(mu_);
  free_.emplace(bytes, p);
}

void DeviceAllocator::free_all() {
  std::lock_guard<std::mutex> g(mu_);
  for (auto& kv : free_) (void)hipFree(kv.second);
  free_.clear();
}

SyncedMemory::~SyncedMemory() {
  if (cpu_ptr_) free(cpu_ptr_);
  if (gpu_ptr_ && own_gpu_)
    Engine::get().dalloc.release(gpu_ptr_, bytes_);
}

void SyncedMemory::to_cpu() {
  switch (head_) {
    case UNINIT:
      cpu_ptr_ = calloc(1, bytes_);
      CHECK_(cpu_ptr_);
      head_ = AT_CPU;
      break;
    case AT_GPU: {
      if (!cpu_ptr_) {
        cpu_ptr_ = malloc(bytes_);
        CHECK_(cpu_ptr_);
      }
      Engine& E = Engine::get();
      HIP_CHECK(hipMemcpyAsync(cpu_ptr_, gpu_ptr_, bytes_,
                               hipMemcpyDeviceToHost, E.stream));
      HIP_CHECK(hipStreamSynchronize(E.stream));
      head_ = SYNCED;
      break;
    }
    default:
      break;
  }
}

void SyncedMemory::to_gpu() {
  Engine& E = Engine::get();
  CHECK_(E.mode == Mode::GPU) << "gpu_data requested in CPU mode";
  switch (head_) {
    case UNINIT:
      gpu_ptr_ = E.dalloc.alloc(bytes_);
      own_gpu_ = true;
      HIP_CHECK(hipMemsetAsync(gpu_ptr_, 0, bytes_, E.stream));
      head_ = AT_GPU;
      break;
    case AT_CPU: {
      if (!gpu_ptr_) {
        gpu_ptr_ = E.dalloc.alloc(bytes_);
        own_gpu_ = true;
      }
      HIP_CHECK(hipMemcpyAsync(gpu_ptr_, cpu_ptr_, bytes_,
                               hipMemcpyHostToDevice, E.stream));
      HIP_CHECK(hipStreamSynchronize(E.stream));
      head_ = SYNCED;
      break;
    }
    default:
      break;
  }
}

const void* SyncedMemory::cpu_data() {
  to_cpu();
  return cpu_ptr_;
}
const void* SyncedMemory::gpu_data() {
  to_gpu();
  return gpu_ptr_;
}
void* SyncedMemory::mutable_cpu_data() {
  to_cpu();
  head_ = AT_CPU;
  return cpu_ptr_;
}
void* SyncedMemory::mutable_gpu_data() {
  to_gpu();
  head_ = AT_GPU;
  return gpu_ptr_;
}

void SyncedMemory::set_gpu_view(void* p) {
  if (gpu_ptr_ && own_gpu_) Engine::get().dalloc.release(gpu_ptr_, bytes_);
  gpu_ptr_ = p;
  own_gpu_ = false;
  head_ = AT_GPU;
}

void Blob::Reshape(const std::vector<int>& shape) {
  shape_ = shape;
  long c = 1;
  for (int d : shape) {
    CHECK_GE_(d, 0);
    c *= d;
  }
  count_ = c;
  const size_t bytes = padded(c) * sizeof(float);
  if (!data_ || data_->size() < bytes) data_.reset(new SyncedMemory(bytes));
  if (!diff_ || diff_->size() < bytes) diff_.reset(new SyncedMemory(bytes));
}

void Blob::set_data_const(float v) {
  float* p = mutable_cpu_data();
  for (long i = 0; i < count_; ++i) p[i] = v;
}
void Blob::set_diff_const(float v) {
  float* p = mutable_cpu_diff();
  for (long i = 0; i < count_; ++i) p[i] = v;
}

}  // namespace camd
