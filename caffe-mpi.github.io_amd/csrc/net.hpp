// net.hpp — prototxt-defined layer graph with topological execution and the
// backward→reduce producer hook (reference src/caffe/net.cpp: Init :64-405,
// ForwardFromTo :669, BackwardFromToAu :722-751, InitializeLearnableDiffSpace
// :1350-1374).  MI355X re-design notes:
//  - one contiguous device arena for learnable diffs (and a parallel arena
//    for SGD history), laid out in BACKWARD COMPLETION order so the bucketed
//    all-reduce flushes flat ascending ranges;
//  - instead of the reference's reduce thread + blocking queue
//    (net.cpp:757-877), backward records a hipEvent per layer and hands
//    (param_id, event) to the reducer, which chains the collective and the
//    fused update on the side comm stream — same overlap, no host threads.
#pragma once

#include "layers.hpp"

namespace camd {

struct ReduceHook {
  // called right after the producing layer's backward kernels are launched;
  // `done` is recorded on the compute stream (null in CPU mode)
  virtual void param_ready(int param_id, hipEvent_t done) = 0;
  virtual void iteration_end(hipEvent_t backward_done) = 0;
  virtual ~ReduceHook() = default;
};

class Net {
 public:
  Net(const PMsgPtr& net_param, Phase phase, int batch_override = 0);

  void Forward();
  void Backward(ReduceHook* hook = nullptr);
  float loss();  // syncs; Σ loss_weight · loss-top
  // test-score outputs: every loss-weighted top + every Accuracy top
  std::vector<std::pair<std::string, float>> scores();

  struct LParam {
    Blob* blob;
    Layer* layer;
    int blob_idx;
    float lr_mult, decay_mult;
    long offset;  // elements into the diff arena (GPU mode), padded
    long count;
  };
  std::vector<LParam>& learnable_params() { return params_; }
  long learnable_count() const { return arena_count_; }
  // params_ is arena (reverse-layer) order; the reference serializes SGD
  // history in FORWARD learnable-param order (sgd_solver.cpp:262-353) —
  // snapshot/restore must walk this permutation for wire compatibility
  std::vector<int> forward_param_order() const {
    std::map<const Layer*, int> pos;
    for (size_t i = 0; i < layers_.size(); ++i) pos[layers_[i].get()] = (int)i;
    std::vector<int> idx(params_.size());
    for (size_t i = 0; i < idx.size(); ++i) idx[i] = (int)i;
    std::stable_sort(idx.begin(), idx.end(), [&](int a, int b) {
      const int la = pos.at(params_[a].layer), lb = pos.at(params_[b].layer);
      if (la != lb) return la < lb;
      return params_[a].blob_idx < params_[b].blob_idx;
    });
    return idx;
  }
  float* diff_arena() { return diff_arena_; }
  float* data_arena() { return nullptr; }  // weights stay per-blob this round

  const std::vector<std::shared_ptr<Layer>>& layers() const {
    return layers_;
  }
  Blob* blob_by_name(const std::string& n) {
    auto it = blob_map_.find(n);
    return it == blob_map_.end() ? nullptr : it->second.get();
  }
  std::vector<std::string> blob_names() const;
  const std::string& name() const { return name_; }
  Phase phase() const { return phase_; }

  // copy weights from another net (test-net sharing / snapshot restore)
  void ShareTrainedLayersWith(Net& other);

  // .caffemodel binaryproto interop (reference Net weight load/save,
  // net.cpp:1055-1248; format: proto_wire.hpp)
  void SaveWeights(const std::string& path);
  void LoadWeights(const std::string& path);

  // `caffe time`-style per-layer forward/backward timing report
  void time_layers(int iters);

 private:
  void init(const PMsgPtr& msg, int batch_override);
  void insert_splits(std::vector<PMsgPtr>& layer_msgs);
  void setup_arena();

  std::string name_;
  Phase phase_;
  std::vector<std::shared_ptr<Layer>> layers_;
  std::vector<std::vector<Blob*>> bottoms_, tops_;
  std::vector<std::vector<bool>> prop_down_;
  std::vector<bool> layer_need_bwd_;
  std::map<std::string, std::shared_ptr<Blob>> blob_map_;
  std::vector<LParam> params_;
  long arena_count_ = 0;
  float* diff_arena_ = nullptr;
  std::vector<hipEvent_t> layer_events_;
};

}  // namespace camd
