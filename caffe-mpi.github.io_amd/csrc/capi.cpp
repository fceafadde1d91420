// capi.cpp — C-ABI implementation (see include/caffe_amd.h for the mapping
// to reference interfaces).
#include "../../include/caffe_amd.h"
#include "solver.hpp"

namespace camd {
extern int g_syn_shape[3];  // defined in layers_cpu.cpp
}

using namespace camd;

namespace {
thread_local std::string g_err;
std::vector<std::shared_ptr<Solver>> g_solvers;
std::vector<std::shared_ptr<Net>> g_nets;

#define API_TRY try {
#define API_CATCH                      \
  }                                    \
  catch (const std::exception& e) {    \
    g_err = e.what();                  \
    return -1;                         \
  }
#define API_CATCH_NULL                 \
  }                                    \
  catch (const std::exception& e) {    \
    g_err = e.what();                  \
    return nullptr;                    \
  }

Solver* S(caffe_solver_t s) { return (Solver*)s; }
Net* N(caffe_net_t n) { return (Net*)n; }
}  // namespace

extern "C" {

const char* caffe_last_error(void) { return g_err.c_str(); }

int caffe_set_mode(int mode, int device) {
  API_TRY
  if (mode == 1)
    Engine::get().set_mode_gpu(device);
  else
    Engine::get().mode = Mode::CPU;
  return 0;
  API_CATCH
}

// multi-node uid bootstrap (csrc/bootstrap.cpp) — exposed so the TCP
// exchange is testable without GPUs or multiple boxes
int caffe_uid_serve(const uint8_t* uid, int port, int nclients) {
  API_TRY
  return uid_serve(uid, 128, port, nclients);
  API_CATCH
}
int caffe_uid_fetch(uint8_t* out, const char* host, int port,
                    int timeout_s) {
  API_TRY
  return uid_fetch(out, 128, host, port, timeout_s);
  API_CATCH
}

// data-stream iteration counter (the LMDB cursor / synthetic stream
// position; Solver::Step drives it during training — tests reset it)
int caffe_set_data_iter(uint64_t iter) {
  API_TRY
  Engine::get().data_iter = iter;
  return 0;
  API_CATCH
}

// rank/world without a communicator (single-process tests of the
// rank-sharded data feed; comm_init sets the same fields when a real
// collective attaches — reference P2PSync rank plumbing)
int caffe_set_rank_world(int rank, int world) {
  API_TRY
  Engine& E = Engine::get();
  E.rank = rank;
  E.world = world;
  E.cpu_rng.seed(E.seed + (uint64_t)rank);
  return 0;
  API_CATCH
}

// GEMM compute dtype: "f32" (exact MFMA, default) or "bf16" (bf16 MFMA,
// fp32 accumulation — mixed precision, the reference Ftype/Btype analog).
// Storage and every non-GEMM op stay fp32.
int caffe_set_compute(const char* dtype) {
  API_TRY
  std::string d = dtype ? dtype : "";
  if (d == "f32" || d == "fp32" || d == "float")
    Engine::get().gemm_bf16 = false;
  else if (d == "bf16" || d == "bfloat16")
    Engine::get().gemm_bf16 = true;
  else
    CAMD_FATAL << "unknown compute dtype " << d;
  return 0;
  API_CATCH
}

int caffe_set_random_seed(uint64_t seed) {
  API_TRY
  Engine& E = Engine::get();
  E.seed = seed;
  E.cpu_rng.seed(seed + (uint64_t)E.rank);
  return 0;
  API_CATCH
}

int caffe_set_synthetic_shape(int c, int h, int w, int num_classes) {
  API_TRY
  camd::g_syn_shape[0] = c;
  camd::g_syn_shape[1] = h;
  camd::g_syn_shape[2] = w;
  if (num_classes > 0) Engine::get().syn_classes = num_classes;
  return 0;
  API_CATCH
}

int caffe_set_perf_timing(int enable) {
  Engine::get().perf_timing = enable != 0;
  return 0;
}

caffe_solver_t caffe_solver_create(const char* path, int batch_override) {
  API_TRY
  auto s = create_solver_from_file(path, batch_override);
  g_solvers.push_back(s);
  return (caffe_solver_t)s.get();
  API_CATCH_NULL
}

caffe_solver_t caffe_solver_create_from_text(const char* text,
                                             int batch_override) {
  API_TRY
  auto s = std::make_shared<Solver>(parse_prototxt(text), batch_override);
  g_solvers.push_back(s);
  return (caffe_solver_t)s.get();
  API_CATCH_NULL
}

void caffe_solver_free(caffe_solver_t s) {
  for (auto it = g_solvers.begin(); it != g_solvers.end(); ++it)
    if (it->get() == s) {
      g_solvers.erase(it);
      return;
    }
}

int caffe_solver_step(caffe_solver_t s, int iters) {
  API_TRY
  S(s)->Step(iters);
  return 0;
  API_CATCH
}

long caffe_solver_iter(caffe_solver_t s) { return S(s)->iter(); }

float caffe_solver_loss(caffe_solver_t s) {
  try {
    return S(s)->last_loss();
  } catch (const std::exception& e) {
    g_err = e.what();
    return -1.f;
  }
}

caffe_net_t caffe_solver_net(caffe_solver_t s) {
  return (caffe_net_t)&S(s)->net();
}

int caffe_solver_snapshot(caffe_solver_t s) {
  API_TRY
  S(s)->Snapshot();
  return 0;
  API_CATCH
}

int caffe_solver_restore(caffe_solver_t s, const char* state_path) {
  API_TRY
  S(s)->Restore(state_path);
  return 0;
  API_CATCH
}

int caffe_net_save_weights(caffe_net_t n, const char* path) {
  API_TRY
  N(n)->SaveWeights(path);
  return 0;
  API_CATCH
}

int caffe_net_load_weights(caffe_net_t n, const char* path) {
  API_TRY
  N(n)->LoadWeights(path);
  return 0;
  API_CATCH
}

int caffe_comm_selftest(void) {
  API_TRY
  return rccl_selftest();
  API_CATCH
}

int caffe_comm_unique_id(uint8_t out[128]) {
  API_TRY
  rccl_unique_id(out);
  return 0;
  API_CATCH
}

int caffe_comm_init(caffe_solver_t s, int rank, int world,
                    const uint8_t id[128]) {
  API_TRY
  Engine& E = Engine::get();
  E.rank = rank;
  E.world = world;
  E.cpu_rng.seed(E.seed + (uint64_t)rank);
  S(s)->set_comm(make_rccl_comm(rank, world, id));
  return 0;
  API_CATCH
}

int caffe_comm_bcast_weights(caffe_solver_t s) {
  API_TRY
  S(s)->bcast_weights();
  return 0;
  API_CATCH
}

int caffe_comm_set_callback(caffe_solver_t s, caffe_allreduce_cb cb,
                            void* ud, int world) {
  API_TRY
  Engine::get().world = world;
  S(s)->set_comm(make_callback_comm(cb, ud, world));
  return 0;
  API_CATCH
}

caffe_net_t caffe_net_create(const char* path, int phase,
                             int batch_override) {
  API_TRY
  auto n = std::make_shared<Net>(parse_prototxt_file(path),
                                 phase ? Phase::TEST : Phase::TRAIN,
                                 batch_override);
  g_nets.push_back(n);
  return (caffe_net_t)n.get();
  API_CATCH_NULL
}

void caffe_net_free(caffe_net_t n) {
  for (auto it = g_nets.begin(); it != g_nets.end(); ++it)
    if (it->get() == n) {
      g_nets.erase(it);
      return;
    }
}

int caffe_net_forward(caffe_net_t n) {
  API_TRY
  N(n)->Forward();
  return 0;
  API_CATCH
}

int caffe_net_backward(caffe_net_t n) {
  API_TRY
  N(n)->Backward(nullptr);
  return 0;
  API_CATCH
}

float caffe_net_loss(caffe_net_t n) {
  try {
    return N(n)->loss();
  } catch (const std::exception& e) {
    g_err = e.what();
    return -1.f;
  }
}

int caffe_net_blob_shape(caffe_net_t n, const char* name, int* shape_out,
                         int max_dims, int* ndims_out) {
  API_TRY
  Blob* b = N(n)->blob_by_name(name);
  CHECK_(b) << "no blob " << name;
  const auto& s = b->shape();
  *ndims_out = (int)s.size();
  for (int i = 0; i < (int)s.size() && i < max_dims; ++i) shape_out[i] = s[i];
  return 0;
  API_CATCH
}

int caffe_net_blob_get(caffe_net_t n, const char* name, int diff, float* out,
                       long count) {
  API_TRY
  Blob* b = N(n)->blob_by_name(name);
  CHECK_(b) << "no blob " << name;
  CHECK_LE_(count, b->count());
  Engine::get().sync();
  const float* src = diff ? b->cpu_diff() : b->cpu_data();
  memcpy(out, src, sizeof(float) * count);
  return 0;
  API_CATCH
}

int caffe_net_blob_set(caffe_net_t n, const char* name, int diff,
                       const float* in, long count) {
  API_TRY
  Blob* b = N(n)->blob_by_name(name);
  CHECK_(b) << "no blob " << name;
  CHECK_LE_(count, b->count());
  float* dst = diff ? b->mutable_cpu_diff() : b->mutable_cpu_data();
  memcpy(dst, in, sizeof(float) * count);
  return 0;
  API_CATCH
}

// ---- pycaffe-shim surface (round 2): zero-copy CPU pointers with the
// SyncedMemory head semantics pycaffe relied on (mutable access marks the
// host copy dirty; the next GPU use re-syncs), plus layer enumeration so
// `net.blobs` / `net.params` / `net.layers` can be dict-shaped.
float* caffe_net_blob_cpu_ptr(caffe_net_t n, const char* name, int diff,
                              int writable) {
  try {
    Blob* b = N(n)->blob_by_name(name);
    CHECK_(b) << "no blob " << name;
    Engine::get().sync();
    if (writable)
      return diff ? b->mutable_cpu_diff() : b->mutable_cpu_data();
    return const_cast<float*>(diff ? b->cpu_diff() : b->cpu_data());
  } catch (const std::exception& e) {
    g_err = e.what();
    return nullptr;
  }
}

int caffe_net_num_layers(caffe_net_t n) {
  return (int)N(n)->layers().size();
}

// newline-joined blob names in topological order
int caffe_net_blob_names(caffe_net_t n, char* out, int cap) {
  API_TRY
  std::string joined;
  for (auto& nm : N(n)->blob_names()) {
    if (!joined.empty()) joined += "\n";
    joined += nm;
  }
  snprintf(out, cap, "%s", joined.c_str());
  return 0;
  API_CATCH
}

int caffe_net_layer_info(caffe_net_t n, int idx, char* name_out,
                         int name_cap, char* type_out, int type_cap,
                         int* num_blobs_out) {
  API_TRY
  auto& ls = N(n)->layers();
  CHECK_LT_(idx, (int)ls.size());
  snprintf(name_out, name_cap, "%s", ls[idx]->name().c_str());
  snprintf(type_out, type_cap, "%s", ls[idx]->type().c_str());
  *num_blobs_out = (int)ls[idx]->blobs().size();
  return 0;
  API_CATCH
}

static Layer* find_layer(caffe_net_t n, const char* lname) {
  for (auto& l : N(n)->layers())
    if (l->name() == lname) return l.get();
  CAMD_FATAL << "no layer " << lname;
}

int caffe_net_layer_blob_shape(caffe_net_t n, const char* lname, int bidx,
                               int* shape_out, int max_dims,
                               int* ndims_out) {
  API_TRY
  Layer* l = find_layer(n, lname);
  CHECK_LT_(bidx, (int)l->blobs().size());
  const auto& s = l->blobs()[bidx]->shape();
  *ndims_out = (int)s.size();
  for (int i = 0; i < (int)s.size() && i < max_dims; ++i)
    shape_out[i] = s[i];
  return 0;
  API_CATCH
}

float* caffe_net_layer_blob_cpu_ptr(caffe_net_t n, const char* lname,
                                    int bidx, int diff, int writable) {
  try {
    Layer* l = find_layer(n, lname);
    CHECK_LT_(bidx, (int)l->blobs().size());
    Blob* b = l->blobs()[bidx].get();
    Engine::get().sync();
    if (writable)
      return diff ? b->mutable_cpu_diff() : b->mutable_cpu_data();
    return const_cast<float*>(diff ? b->cpu_diff() : b->cpu_data());
  } catch (const std::exception& e) {
    g_err = e.what();
    return nullptr;
  }
}

int caffe_net_num_params(caffe_net_t n) {
  return (int)N(n)->learnable_params().size();
}

int caffe_net_param_info(caffe_net_t n, int idx, char* name_out,
                         int name_cap, int* blob_idx_out, long* count_out) {
  API_TRY
  auto& ps = N(n)->learnable_params();
  CHECK_LT_(idx, (int)ps.size());
  snprintf(name_out, name_cap, "%s", ps[idx].layer->name().c_str());
  *blob_idx_out = ps[idx].blob_idx;
  *count_out = ps[idx].count;
  return 0;
  API_CATCH
}

int caffe_net_param_get(caffe_net_t n, int idx, int diff, float* out,
                        long count) {
  API_TRY
  auto& ps = N(n)->learnable_params();
  CHECK_LT_(idx, (int)ps.size());
  Blob* b = ps[idx].blob;
  CHECK_LE_(count, b->count());
  Engine::get().sync();
  memcpy(out, diff ? b->cpu_diff() : b->cpu_data(), sizeof(float) * count);
  return 0;
  API_CATCH
}

int caffe_net_param_set(caffe_net_t n, int idx, const float* in, long count) {
  API_TRY
  auto& ps = N(n)->learnable_params();
  CHECK_LT_(idx, (int)ps.size());
  Blob* b = ps[idx].blob;
  CHECK_LE_(count, b->count());
  memcpy(b->mutable_cpu_data(), in, sizeof(float) * count);
  return 0;
  API_CATCH
}

int caffe_device_synchronize(void) {
  API_TRY
  Engine::get().sync();
  Engine::get().drain_events();
  return 0;
  API_CATCH
}

int caffe_perf_snapshot(char* names, int name_cap, long* launches,
                        double* flops, double* bytes, double* ns,
                        int max_rows) {
  API_TRY
  Engine& E = Engine::get();
  int row = 0;
  for (auto& kv : E.perf()) {
    if (row >= max_rows) break;
    snprintf(names + (size_t)row * name_cap, name_cap, "%s",
             kv.first.c_str());
    launches[row] = kv.second.launches.load();
    flops[row] = kv.second.flops.load();
    bytes[row] = kv.second.bytes.load();
    ns[row] = kv.second.ns.load();
    ++row;
  }
  return row;
  API_CATCH
}

int caffe_perf_reset(void) {
  for (auto& kv : Engine::get().perf()) {
    kv.second.launches = 0;
    kv.second.flops = 0;
    kv.second.bytes = 0;
    kv.second.ns = 0;
  }
  return 0;
}

}  // extern "C"
