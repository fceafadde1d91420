#include "solver.hpp"

#include <chrono>
#include <cmath>
#include <filesystem>
#include <fstream>

#include "proto_wire.hpp"

namespace camd {

Solver::Solver(const PMsgPtr& sp, int batch_override) : param_(sp) {
  PMsgPtr net_msg;
  if (sp->has("net")) {
    net_msg = parse_prototxt_file(sp->str("net"));
  } else if (sp->sub("net_param")) {
    net_msg = sp->sub("net_param");
  } else {
    CAMD_FATAL << "solver has no net";
  }
  net_msg_ = net_msg;
  // solver-pinned seed (models use random_seed; `caffe time` pins 1371)
  Engine& E = Engine::get();
  if (sp->has("random_seed")) {
    E.seed = (uint64_t)sp->inum("random_seed");
    E.cpu_rng.seed(E.seed + (uint64_t)E.rank);  // +rank: parallel.cpp:179-187
  }
  net_.reset(new Net(net_msg, Phase::TRAIN, batch_override));
  weight_decay_ = (float)sp->num("weight_decay", 0.0);
  const long nparams = (long)net_->learnable_params().size();
  const long buckets = sp->inum("reduce_buckets", 6);  // caffe.proto:140
  bucket_budget_ =
      std::max<long>(1, net_->learnable_count() / std::max<long>(1, buckets));
  if (E.mode == Mode::GPU && net_->learnable_count() > 0) {
    history_ =
        (float*)E.dalloc.alloc(sizeof(float) * net_->learnable_count());
    HIP_CHECK(hipMemsetAsync(history_, 0,
                             sizeof(float) * net_->learnable_count(),
                             E.stream));
  } else if (nparams > 0) {
    host_history_.assign(net_->learnable_count(), 0.f);
  }
}

Solver::~Solver() {
  if (history_)
    Engine::get().dalloc.release(history_,
                                 sizeof(float) * net_->learnable_count());
  if (acc_)
    Engine::get().dalloc.release(acc_,
                                 sizeof(float) * net_->learnable_count());
}

// acc += diff (merge_back=false, after each non-final sub-pass) or
// diff += acc then zero acc (merge_back=true, after the final sub-pass)
void Solver::accumulate_diffs(bool merge_back) {
  Engine& E = Engine::get();
  const long total = net_->learnable_count();
  if (E.mode == Mode::GPU) {
    if (!acc_) {
      acc_ = (float*)E.dalloc.alloc(sizeof(float) * total);
      HIP_CHECK(hipMemsetAsync(acc_, 0, sizeof(float) * total, E.stream));
    }
    if (!merge_back) {
      gpu::axpy(E.stream, total, 1.f, net_->diff_arena(), acc_);
    } else {
      gpu::axpy(E.stream, total, 1.f, acc_, net_->diff_arena());
      HIP_CHECK(hipMemsetAsync(acc_, 0, sizeof(float) * total, E.stream));
    }
  } else {
    if (host_acc_.empty()) host_acc_.assign(total, 0.f);
    for (auto& p : net_->learnable_params()) {
      float* diff = p.blob->mutable_cpu_diff();
      float* acc = host_acc_.data() + p.offset;
      if (!merge_back)
        cpu::axpy(p.count, 1.f, diff, acc);
      else
        cpu::axpy(p.count, 1.f, acc, diff);
    }
    if (merge_back) std::fill(host_acc_.begin(), host_acc_.end(), 0.f);
  }
}

float Solver::GetLearningRate() const {
  // sgd_solver.cpp:24-66 (+ rampup :27-33)
  const double base_lr = param_->num("base_lr");
  const long rampup = param_->inum("rampup_interval", 0);
  if (iter_ < rampup) {
    const double r0 = param_->num("rampup_lr", 0.0);
    return (float)(r0 + (base_lr - r0) * ((double)iter_ / rampup));
  }
  const std::string policy = param_->str("lr_policy", "fixed");
  if (policy == "fixed") return (float)base_lr;
  if (policy == "step") {
    const long step = iter_ / std::max<long>(1, param_->inum("stepsize", 1));
    return (float)(base_lr * std::pow(param_->num("gamma", 0.1), step));
  }
  if (policy == "exp")
    return (float)(base_lr * std::pow(param_->num("gamma", 0.1), iter_));
  if (policy == "inv")
    return (float)(base_lr *
                   std::pow(1.0 + param_->num("gamma") * iter_,
                            -param_->num("power")));
  if (policy == "multistep") {
    auto sv = param_->inums("stepvalue");
    while (current_step_ < (int)sv.size() && iter_ >= sv[current_step_])
      ++current_step_;
    return (float)(base_lr *
                   std::pow(param_->num("gamma", 0.1), current_step_));
  }
  if (policy == "poly") {
    const double min_lr = param_->num("min_lr", 0.0);
    return (float)(min_lr +
                   (base_lr - min_lr) *
                       std::pow(1.0 - (double)iter_ /
                                          param_->num("max_iter", 1),
                                param_->num("power")));
  }
  if (policy == "sigmoid")
    return (float)(base_lr /
                   (1.0 + std::exp(-param_->num("gamma") *
                                   (double)(iter_ -
                                            param_->inum("stepsize")))));
  CAMD_FATAL << "unknown lr_policy " << policy;
}

void Solver::bcast_weights() {
  if (!comm_ || comm_->world() < 2) return;
  Engine& E = Engine::get();
  for (auto& p : net_->learnable_params()) {
    float* ptr = E.mode == Mode::GPU ? p.blob->mutable_gpu_data()
                                     : p.blob->mutable_cpu_data();
    comm_->bcast(ptr, p.count, 0, E.stream);
  }
  E.sync();
}

void Reducer::start_iteration() {
  bucket_start_ = bucket_end_ = 0;
}

void Reducer::param_ready(int param_id, hipEvent_t done) {
  auto& params = solver_->net().learnable_params();
  CHECK_EQ_((long)param_id, bucket_end_) << "params must arrive in order";
  bucket_end_ = param_id + 1;
  const auto& first = params[bucket_start_];
  const auto& last = params[bucket_end_ - 1];
  const long span = last.offset + (long)padded(last.count) - first.offset;
  if (span >= solver_->bucket_budget_) flush(done);
}

void Solver::ensure_seg_table() {
  // the table holds per-param lr/decay MULTIPLIERS (immutable after net
  // build) — per-iteration lr/decay are kernel arguments, so the table is
  // uploaded exactly once, synchronously (no pageable-async lifetime
  // hazard, no H2D copies in the iteration loop)
  if (d_seg_off_) return;
  Engine& E = Engine::get();
  auto& params = net_->learnable_params();
  const int n = (int)params.size();
  std::vector<long> offs(n);
  std::vector<float*> wps(n);
  std::vector<float> lrms(n), dcms(n);
  for (int i = 0; i < n; ++i) {
    offs[i] = params[i].offset;
    wps[i] = params[i].blob->mutable_gpu_data();
    lrms[i] = params[i].lr_mult;
    dcms[i] = params[i].decay_mult;
  }
  d_seg_off_ = (long*)E.dalloc.alloc(n * sizeof(long));
  d_w_ptrs_ = (float**)E.dalloc.alloc(n * sizeof(float*));
  d_lrs_ = (float*)E.dalloc.alloc(n * sizeof(float));
  d_decays_ = (float*)E.dalloc.alloc(n * sizeof(float));
  HIP_CHECK(hipMemcpy(d_seg_off_, offs.data(), n * sizeof(long),
                      hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(d_w_ptrs_, wps.data(), n * sizeof(float*),
                      hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(d_lrs_, lrms.data(), n * sizeof(float),
                      hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(d_decays_, dcms.data(), n * sizeof(float),
                      hipMemcpyHostToDevice));
}

void Reducer::flush(hipEvent_t ev) {
  if (bucket_end_ == bucket_start_) return;
  Solver& S = *solver_;
  Engine& E = Engine::get();
  auto& params = S.net().learnable_params();
  const auto& first = params[bucket_start_];
  const auto& last = params[bucket_end_ - 1];
  const long count = last.offset + (long)padded(last.count) - first.offset;
  if (E.mode == Mode::GPU) {
    if (ev) HIP_CHECK(hipStreamWaitEvent(E.comm_stream, ev, 0));
    if (S.comm_ && S.comm_->world() > 1) {
      PerfScope ps("allreduce", E.comm_stream, 0, 2.0 * count * 4);
      S.comm_->allreduce(S.net().diff_arena() + first.offset, count,
                         E.comm_stream);
    }
    // bucket-fused update: one launch over the flat arena range
    S.ensure_seg_table();
    gpu::sgd_update_segmented(
        E.comm_stream, first.offset, first.offset + count,
        S.net().diff_arena(), S.history(), S.d_seg_off_, S.d_w_ptrs_,
        S.d_lrs_, S.d_decays_, (int)params.size(), S.cur_mom_, S.cur_lr_,
        S.weight_decay_, S.grad_scale_);
  } else {
    if (S.comm_ && S.comm_->world() > 1) {
      // CPU buckets: gather the param diffs into one flat range is
      // unnecessary — host diffs are per-blob; reduce each param
      for (long k = bucket_start_; k < bucket_end_; ++k)
        S.comm_->allreduce(params[k].blob->mutable_cpu_diff(),
                           params[k].count, nullptr);
    }
    for (long k = bucket_start_; k < bucket_end_; ++k) {
      const auto& p = params[k];
      float* g = p.blob->mutable_cpu_diff();
      float* w = p.blob->mutable_cpu_data();
      float* h = S.host_history().data() + p.offset;
      const float lr = S.cur_lr_ * p.lr_mult;
      const float decay = S.weight_decay_ * p.decay_mult;
      for (long i = 0; i < p.count; ++i) {
        float gi = g[i] * S.grad_scale_ + decay * w[i];
        gi = h[i] = S.cur_mom_ * h[i] + lr * gi;
        w[i] -= gi;
        g[i] = 0.f;
      }
    }
  }
  bucket_start_ = bucket_end_;
}

void Reducer::iteration_end(hipEvent_t backward_done) {
  flush(backward_done);
  Engine& E = Engine::get();
  if (E.mode == Mode::GPU) {
    if (!comm_done_ev_)
      HIP_CHECK(
          hipEventCreateWithFlags(&comm_done_ev_, hipEventDisableTiming));
    HIP_CHECK(hipEventRecord(comm_done_ev_, E.comm_stream));
    // next iteration's forward must see the updated weights
    HIP_CHECK(hipStreamWaitEvent(E.stream, comm_done_ev_, 0));
  }
}

void Solver::Step(int iters) {
  Engine& E = Engine::get();
  using clock = std::chrono::steady_clock;
  const long display = param_->inum("display", 0);
  const long test_interval = param_->inum("test_interval", 0);
  const long test_iter = param_->inum("test_iter", 1);
  const long snap_interval = param_->inum("snapshot", 0);
  // gradient accumulation: iter_size fwd/bwd passes per update, diffs
  // accumulate in the arena, one reduce+update scaled by 1/iter_size
  // (reference solver.cpp:279-297 Step loop + SGDSolver::Normalize)
  const long iter_size = std::max<long>(1, param_->inum("iter_size", 1));
  // this fork always runs a 1-iter test at iter 0 regardless of
  // test_initialization (reference solver.cpp:243-248)
  if (iter_ == 0 && test_interval > 0 && test_net()) TestAll(1);
  // reference-style perf accounting (solver.cpp:299 skips warmup): the
  // first Step call (warmup/compile) is excluded; later calls are timed at
  // call granularity with a single sync so the launch-ahead pipeline stays
  // intact (no per-iteration host synchronization in this engine)
  const bool timed = skipped_iters_ > 0;
  std::chrono::time_point<clock> tstep0;
  if (timed) {
    E.sync();
    tstep0 = clock::now();
  }
  int done_iters = 0;
  for (int i = 0; i < iters; ++i) {
    if (action_fn_) {
      const SolverAction a = action_fn_();
      if (a == SolverAction::SNAPSHOT) {
        if (snapshot_enabled_) Snapshot();
      } else if (a == SolverAction::STOP) {
        early_exit_ = true;
        break;
      }
    }
    if (test_interval > 0 && iter_ > 0 && iter_ % test_interval == 0 &&
        test_net())
      TestAll(test_iter);
    cur_lr_ = GetLearningRate();
    cur_mom_ = GetMomentum();
    const float world =
        comm_ && comm_->world() > 1 ? (float)comm_->world() : 1.f;
    grad_scale_ = 1.f / (world * (float)iter_size);
    reducer_.start_iteration();
    if (iter_size == 1) {
      E.data_iter = (uint64_t)iter_;
      net_->Forward();
      net_->Backward(&reducer_);
    } else {
      // layer backwards overwrite their param diffs, so the accumulation
      // runs hookless with a side buffer; the reducer is driven once at
      // the end (overlap is moot — sub-passes serialize by construction)
      for (long sub = 0; sub < iter_size; ++sub) {
        // sub-iteration granularity: each pass sees a fresh synthetic batch
        E.data_iter = (uint64_t)(iter_ * iter_size + sub);
        net_->Forward();
        net_->Backward(nullptr);
        if (sub + 1 < iter_size) accumulate_diffs(false);
      }
      accumulate_diffs(true);
      E.sync();  // diffs final on E.stream before comm_stream consumes them
      const int nparams = (int)net_->learnable_params().size();
      for (int k = 0; k < nparams; ++k) reducer_.param_ready(k, nullptr);
      reducer_.iteration_end(nullptr);
    }
    ++iter_;
    ++done_iters;
    if (snap_interval > 0 && iter_ % snap_interval == 0 &&
        snapshot_enabled_)
      Snapshot();
    if (display > 0 && iter_ % display == 0) {
      const float l = net_->loss();
      fprintf(stderr, "[caffe_amd] Iteration %ld, loss = %g, lr = %g\n",
              iter_, l, cur_lr_);
    }
  }
  if (timed) {
    E.sync();
    perf_seconds_ +=
        std::chrono::duration<double>(clock::now() - tstep0).count();
    perf_iters_ += done_iters;
  } else {
    skipped_iters_ += done_iters;
  }
}

std::shared_ptr<Solver> create_solver_from_file(const std::string& path,
                                                int batch_override) {
  return std::make_shared<Solver>(parse_prototxt_file(path), batch_override);
}

double Solver::perf_img_per_sec() const {
  if (perf_iters_ == 0 || perf_seconds_ <= 0) return 0.0;
  long batch = 0;
  for (auto& l : net_->layers())
    if (auto* d = dynamic_cast<const DataLayer*>(l.get())) {
      batch = d->batch_;
      break;
    }
  return perf_iters_ / perf_seconds_ * batch;
}

void Solver::print_perf_report() const {
  if (perf_iters_ == 0 || perf_seconds_ <= 0) return;
  // batch size from the train net's data layer (first layer top 0)
  long batch = 0;
  for (auto& l : net_->layers())
    if (auto* d = dynamic_cast<const DataLayer*>(l.get())) {
      batch = d->batch_;
      break;
    }
  const double ratio = perf_iters_ / perf_seconds_;
  fprintf(stderr,
          "Solver performance on device %d: %.4g * %ld = %.5g img/sec\n",
          Engine::get().device, ratio, batch, ratio * batch);
}

Net* Solver::test_net() {
  if (test_net_) return test_net_.get();
  // only build when the net defines TEST-phase layers
  bool has_test = false;
  for (auto& lm : net_msg_->subs("layer")) {
    for (auto& inc : lm->subs("include"))
      if (inc->str("phase") == "TEST") has_test = true;
  }
  if (!has_test) return nullptr;
  test_net_.reset(new Net(net_msg_, Phase::TEST));
  test_net_->ShareTrainedLayersWith(*net_);
  return test_net_.get();
}

void Solver::TestAll(long iters) {
  // reference: the test pass runs on the root solver only (solver.cpp:439;
  // cross-rank SharedScores aggregation is multi-node machinery)
  if (Engine::get().rank != 0) return;
  Net* tn = test_net();
  if (!tn) return;
  // average every loss-weighted / Accuracy top over the test iterations
  // (reference Solver::Test, solver.cpp:439-540; cross-rank SharedScores
  // aggregation is multi-node machinery — single-node here)
  std::map<std::string, double> scores;
  // the data feeds key on Engine::data_iter: advance it per TEST forward
  // so successive test iterations see successive batches (with LMDB this
  // walks the test set in order from record 0 — a fixed eval set; the
  // training stream is unaffected, Step re-pins data_iter each train
  // iteration).  Without this every test iteration repeated ONE batch.
  Engine& E = Engine::get();
  const uint64_t saved_iter = E.data_iter;
  for (long it = 0; it < iters; ++it) {
    E.data_iter = (uint64_t)it;
    tn->Forward();
    for (auto& kv : tn->scores()) scores[kv.first] += kv.second;
  }
  E.data_iter = saved_iter;
  for (auto& kv : scores) {
    const double v = kv.second / iters;
    fprintf(stderr, "[caffe_amd] Test net output: %s = %g\n",
            kv.first.c_str(), v);
    if (!std::isfinite(v))
      fprintf(stderr,
              "[caffe_amd] WARNING: non-finite test score for %s — the net "
              "is diverging (or untrained BN stats at iter 0)\n",
              kv.first.c_str());
  }
}

void Solver::Snapshot() {
  const std::string prefix = param_->str("snapshot_prefix", "snapshot");
  const std::string model =
      prefix + "_iter_" + std::to_string(iter_) + ".caffemodel";
  const std::string state =
      prefix + "_iter_" + std::to_string(iter_) + ".solverstate";
  net_->SaveWeights(model);
  // SolverState { iter=1, learned_net=2, history=3 (BlobProto),
  // current_step=4 } — caffe.proto:303-308
  Engine& E = Engine::get();
  wire::Writer sw;
  sw.vint(1, iter_);
  sw.str(2, model);
  std::vector<float> host;
  // history blobs go out in FORWARD learnable-param order — the order the
  // reference writes (sgd_solver.cpp:262-353); the arena itself is
  // reverse-layer order, so walk the permutation
  auto& aparams = net_->learnable_params();
  for (int pi : net_->forward_param_order()) {
    const auto& p = aparams[pi];
    wire::Writer bw;
    if (E.mode == Mode::GPU) {
      host.resize(p.count);
      HIP_CHECK(hipMemcpy(host.data(), history_ + p.offset,
                          sizeof(float) * p.count, hipMemcpyDeviceToHost));
      bw.packed_floats(5, host.data(), p.count);
    } else {
      bw.packed_floats(5, host_history_.data() + p.offset, p.count);
    }
    wire::Writer shw;
    std::vector<int64_t> dims(p.blob->shape().begin(),
                              p.blob->shape().end());
    shw.packed_i64(1, dims);
    bw.submsg(7, shw.out);
    sw.submsg(3, bw.out);
  }
  sw.vint(4, current_step_);
  {
    const auto dir = std::filesystem::path(state).parent_path();
    if (!dir.empty()) std::filesystem::create_directories(dir);
  }
  std::ofstream f(state, std::ios::binary);
  CHECK_(f.good()) << "cannot write " << state;
  f.write(sw.out.data(), (long)sw.out.size());
  fprintf(stderr, "[caffe_amd] Snapshotting to %s\n", model.c_str());
}

void Solver::Restore(const std::string& path) {
  std::ifstream f(path, std::ios::binary);
  CHECK_(f.good()) << "cannot read " << path;
  std::string buf((std::istreambuf_iterator<char>(f)),
                  std::istreambuf_iterator<char>());
  Engine& E = Engine::get();
  wire::Reader r(buf.data(), buf.size());
  wire::Field fld;
  std::string learned;
  size_t hidx = 0;
  auto& params = net_->learnable_params();
  // the wire carries history in forward learnable-param order (reference
  // sgd_solver.cpp:307-353) — map each blob back to its arena slot
  const std::vector<int> order = net_->forward_param_order();
  while (r.next(&fld)) {
    if (fld.num == 1 && fld.wt == 0) iter_ = (long)fld.vint;
    else if (fld.num == 2 && fld.wt == 2) learned.assign(fld.data, fld.len);
    else if (fld.num == 4 && fld.wt == 0) current_step_ = (int)fld.vint;
    else if (fld.num == 3 && fld.wt == 2) {
      auto b = wire::parse_blob(fld.data, fld.len);
      CHECK_LT_(hidx, order.size());
      const auto& p = params[order[hidx]];
      CHECK_EQ_((long)b.data.size(), p.count);
      if (E.mode == Mode::GPU) {
        HIP_CHECK(hipMemcpy(history_ + p.offset, b.data.data(),
                            sizeof(float) * b.data.size(),
                            hipMemcpyHostToDevice));
      } else {
        memcpy(host_history_.data() + p.offset, b.data.data(),
               sizeof(float) * b.data.size());
      }
      ++hidx;
    }
  }
  if (!learned.empty()) net_->LoadWeights(learned);
  fprintf(stderr, "[caffe_amd] Restored iter %ld from %s\n", iter_,
          path.c_str());
}

}  // namespace camd
