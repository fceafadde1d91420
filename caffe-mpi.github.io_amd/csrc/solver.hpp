// solver.hpp — SGD solver + the bucketed gradient reducer.
//
// Reference semantics: solver.cpp Step :187-353, sgd_solver.cpp lr/momentum
// policies :24-91 and fused update :143-149 / sgd_solver.cu:10-20;
// Net::ReduceAndUpdate bucketing net.cpp:757-877 (reduce_buckets default 6,
// caffe.proto:140).  MI355X re-design: the reduce thread is replaced by
// launch-chaining — each bucket's ncclAllReduce (RCCL over xGMI) and the
// per-param fused SGD updates are enqueued on the side comm stream behind a
// hipEvent recorded after the producing layer's backward, so the collective
// overlaps the rest of backward exactly as the reference's thread did, with
// no host synchronization in the loop.
#pragma once

#include "net.hpp"

namespace camd {

// communicator abstraction: RCCL on GPU, host callback for gloo-based CPU
// tests (tests/test_dist_cpu.py), none for single-GPU
struct Comm {
  virtual void allreduce(float* buf, long count, hipStream_t s) = 0;
  virtual int world() const = 0;
  virtual void bcast(float* buf, long count, int root, hipStream_t s) = 0;
  virtual ~Comm() = default;
};

void rccl_unique_id(void* out_bytes128);
// multi-node uid bootstrap over TCP (csrc/bootstrap.cpp — the reference's
// Clusters/MPI_Bcast replacement, clusters.cpp + parallel.cpp:42-45)
int uid_serve(const void* payload, int len, int port, int nclients);
int uid_fetch(void* out, int len, const char* host, int port,
              int timeout_s);
int rccl_selftest();  // world-1 RCCL linkage/call smoke (GPU mode)
std::unique_ptr<Comm> make_rccl_comm(int rank, int world,
                                     const void* uid_bytes128);
using HostAllreduceFn = void (*)(float*, long, void*);
std::unique_ptr<Comm> make_callback_comm(HostAllreduceFn fn, void* ud,
                                         int world);

class Solver;

// reference SolverAction (include/caffe/solver.hpp + util/signal_handler):
// polled once per iteration inside Step — NONE continue, STOP break out
// (requested_early_exit_), SNAPSHOT snapshot and continue
enum class SolverAction { NONE = 0, STOP = 1, SNAPSHOT = 2 };
using ActionRequestFn = SolverAction (*)();

class Reducer : public ReduceHook {
 public:
  Reducer(Solver* s) : solver_(s) {}
  void start_iteration();
  void param_ready(int param_id, hipEvent_t done) override;
  void iteration_end(hipEvent_t backward_done) override;
  hipEvent_t comm_done() { return comm_done_ev_; }

 private:
  void flush(hipEvent_t ev);
  Solver* solver_;
  long bucket_start_ = 0;   // param index where current bucket starts
  long bucket_end_ = 0;     // one past last ready param
  hipEvent_t comm_done_ev_ = nullptr;
};

class Solver {
 public:
  explicit Solver(const PMsgPtr& solver_param, int batch_override = 0);
  ~Solver();

  void Step(int iters);
  float GetLearningRate() const;  // sgd_solver.cpp:24-66 policies + rampup
  float GetMomentum() const { return (float)param_->num("momentum", 0.0); }

  Net& net() { return *net_; }
  Net* test_net();  // lazily built TEST-phase net sharing train weights
  void TestAll(long iters);  // solver.cpp:439-540 semantics
  void Snapshot();           // .caffemodel + .solverstate (solver.cpp:542)
  // reference per-GPU perf report (solver.cpp:619-628): iters/sec x batch,
  // skipping the first two measured iterations like the reference (:299)
  void print_perf_report() const;
  void Restore(const std::string& state_path);
  void LoadWeights(const std::string& model_path) {
    net_->LoadWeights(model_path);
  }
  long iter() const { return iter_; }
  float last_loss() { return net_->loss(); }
  const PMsgPtr& param() const { return param_; }

  void set_action_request(ActionRequestFn fn) { action_fn_ = fn; }
  bool early_exit() const { return early_exit_; }
  // reference: only the root solver snapshots (rank 0's moving averages /
  // params are the ones persisted) — non-root CLI ranks disable the
  // snapshot-interval and signal-triggered snapshots; direct Snapshot()
  // calls (C ABI) are unaffected
  void set_snapshot_enabled(bool e) { snapshot_enabled_ = e; }
  bool snapshot_enabled() const { return snapshot_enabled_; }
  // measured throughput of the timed Step calls (img/sec), 0 before any
  double perf_img_per_sec() const;

  void set_comm(std::unique_ptr<Comm> c) { comm_ = std::move(c); }
  Comm* comm() { return comm_.get(); }
  float* history() { return history_; }
  std::vector<float>& host_history() { return host_history_; }
  void bcast_weights();  // initial weight broadcast (parallel.cpp:208-227)

  // per-iteration cached coefficients for the fused update
  float cur_lr_ = 0.f, cur_mom_ = 0.f, weight_decay_ = 0.f;
  float grad_scale_ = 1.f;
  long bucket_budget_ = 0;  // elements per bucket

 private:
  PMsgPtr param_;
  PMsgPtr net_msg_;
  std::unique_ptr<Net> net_;
  std::unique_ptr<Net> test_net_;
  std::unique_ptr<Comm> comm_;
  Reducer reducer_{this};
  ActionRequestFn action_fn_ = nullptr;
  bool early_exit_ = false;
  bool snapshot_enabled_ = true;
  long iter_ = 0;
  mutable int current_step_ = 0;
  float* history_ = nullptr;  // device arena, diff-arena layout
  std::vector<float> host_history_;
  // device segment table for the bucket-fused update (GPU mode): arena
  // offsets, per-param weight base pointers, lr/decay (mults folded in at
  // flush time by scaling the per-iteration coefficients)
  long* d_seg_off_ = nullptr;
  float** d_w_ptrs_ = nullptr;
  float* d_lrs_ = nullptr;    // per-param lr MULTIPLIERS (uploaded once)
  float* d_decays_ = nullptr; // per-param decay multipliers
  void ensure_seg_table();
  // iter_size accumulation buffer (diff-arena layout; only allocated when
  // iter_size > 1 — layer backwards overwrite their param diffs, so cross-
  // pass accumulation happens here instead of in every wgrad kernel)
  float* acc_ = nullptr;
  std::vector<float> host_acc_;
  void accumulate_diffs(bool merge_back);
  // perf-report bookkeeping
  double perf_seconds_ = 0.0;
  long perf_iters_ = 0;
  long skipped_iters_ = 0;

  friend class Reducer;
};

std::shared_ptr<Solver> create_solver_from_file(const std::string& path,
                                                int batch_override = 0);

}  // namespace camd
