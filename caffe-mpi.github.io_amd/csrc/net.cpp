#include "net.hpp"

#include <chrono>
#include <set>

namespace camd {

namespace {
// does this layer message run in `phase`? (include/exclude NetStateRule
// subset — only `phase` is used by the four model sets)
bool layer_in_phase(const PMsgPtr& lm, Phase phase) {
  const char* want = phase == Phase::TRAIN ? "TRAIN" : "TEST";
  auto incs = lm->subs("include");
  if (!incs.empty()) {
    for (auto& r : incs)
      if (r->str("phase") == want) return true;
    return false;
  }
  for (auto& r : lm->subs("exclude"))
    if (r->str("phase") == want) return false;
  return true;
}
}  // namespace

Net::Net(const PMsgPtr& net_param, Phase phase, int batch_override)
    : phase_(phase) {
  init(net_param, batch_override);
}

// reference src/caffe/util/insert_splits.cpp behaviour: any blob consumed by
// more than one layer below its producer gets a Split layer so every
// consumer owns a private diff that Split's backward sums.
void Net::insert_splits(std::vector<PMsgPtr>& msgs) {
  // count consumptions of each (versioned) blob name
  std::map<std::string, int> last_producer;  // name -> producing msg idx
  std::map<std::string, int> consumers;      // versioned key -> count
  auto vkey = [&](const std::string& b) {
    return b + "#" + std::to_string(last_producer.count(b)
                                        ? last_producer[b]
                                        : -1);
  };
  std::vector<std::vector<std::string>> bkeys(msgs.size());
  for (size_t i = 0; i < msgs.size(); ++i) {
    for (auto& b : msgs[i]->strs("bottom")) {
      const std::string k = vkey(b);
      consumers[k]++;
      bkeys[i].push_back(k);
    }
    for (auto& t : msgs[i]->strs("top")) last_producer[t] = (int)i;
  }
  // rewire
  std::map<std::string, int> split_next;  // key -> next split-top index
  std::map<std::string, std::string> split_base;
  std::vector<PMsgPtr> out;
  last_producer.clear();
  for (size_t i = 0; i < msgs.size(); ++i) {
    // rewrite bottoms that consume split blobs
    auto& m = msgs[i];
    size_t bi = 0;
    for (auto& f : m->fields) {
      if (f.first != "bottom") continue;
      const std::string k = bkeys[i][bi++];
      auto it = split_base.find(k);
      if (it != split_base.end())
        f.second.scalar =
            it->second + "_split_" + std::to_string(split_next[k]++);
    }
    out.push_back(m);
    // after this layer, split any of its tops with >1 consumers
    for (auto& t : m->strs("top")) {
      last_producer[t] = (int)i;
      const std::string k = t + "#" + std::to_string(i);
      if (consumers[k] > 1) {
        auto sp = std::make_shared<PMsg>();
        auto addf = [&](const std::string& n, const std::string& v) {
          PVal pv;
          pv.kind = PVal::SCALAR;
          pv.scalar = v;
          sp->fields.emplace_back(n, pv);
        };
        addf("name", t + "_" + m->str("name") + "_split");
        addf("type", "Split");
        addf("bottom", t);
        for (int c = 0; c < consumers[k]; ++c)
          addf("top", t + "_split_" + std::to_string(c));
        out.push_back(sp);
        split_base[k] = t;
        split_next[k] = 0;
      }
    }
  }
  msgs = std::move(out);
}

void Net::init(const PMsgPtr& msg, int batch_override) {
  name_ = msg->str("name");
  std::vector<PMsgPtr> lmsgs;
  for (auto& lm : msg->subs("layer"))
    if (layer_in_phase(lm, phase_)) {
      // shallow-clone: insert_splits rewrites bottom names in place, and
      // the source tree is shared with other nets (train/test pair)
      auto copy = std::make_shared<PMsg>(*lm);
      lmsgs.push_back(copy);
    }
  CHECK_(!lmsgs.empty()) << "net has no layers for this phase";
  insert_splits(lmsgs);

  for (auto& lm : lmsgs) {
    auto layer = create_layer(lm);
    layer->set_phase(phase_);
    // wire bottoms
    std::vector<Blob*> bottom, top;
    for (auto& b : lm->strs("bottom")) {
      auto it = blob_map_.find(b);
      CHECK_(it != blob_map_.end())
          << "layer " << layer->name() << ": unknown bottom blob " << b;
      bottom.push_back(it->second.get());
    }
    for (auto& t : lm->strs("top")) {
      auto it = blob_map_.find(t);
      if (it != blob_map_.end()) {
        // in-place: top name equals an existing blob (must be a bottom)
        top.push_back(it->second.get());
      } else {
        auto nb = std::make_shared<Blob>();
        blob_map_[t] = nb;
        top.push_back(nb.get());
      }
    }
    const std::string ctx_s =
        "layer '" + layer->name() + "' (" + layer->type() + ") SetUp";
    error_context() = ctx_s.c_str();
    if (batch_override > 0 && layer->type() == "Data") {
      // per-rank batch override (reference divides the prototxt batch
      // across GPUs, parallel.cpp:284-348)
      layer->SetUp(bottom, top);
      static_cast<DataLayer*>(layer.get())->batch_ = batch_override;
      layer->Reshape(bottom, top);
    } else {
      layer->SetUp(bottom, top);
    }
    error_context() = nullptr;
    // loss weights
    auto lw = lm->nums("loss_weight");
    for (size_t i = 0; i < top.size(); ++i) {
      float w = i < lw.size() ? (float)lw[i]
                              : (i == 0 ? layer->default_loss_weight() : 0.f);
      layer->set_loss((int)i, w);
    }
    layers_.push_back(layer);
    bottoms_.push_back(bottom);
    tops_.push_back(top);
  }

  // need-backward / propagate-down flags (net.cpp Init bottom-up logic;
  // force_backward marks every non-data path like the reference does)
  const bool force = msg->boolean("force_backward", false);
  std::map<Blob*, bool> blob_needs;
  layer_need_bwd_.resize(layers_.size(), false);
  prop_down_.resize(layers_.size());
  for (size_t i = 0; i < layers_.size(); ++i) {
    const bool is_source = layers_[i]->type() == "Data" ||
                           layers_[i]->type() == "Accuracy";
    bool need = !layers_[i]->blobs().empty();
    prop_down_[i].resize(bottoms_[i].size(), false);
    for (size_t b = 0; b < bottoms_[i].size(); ++b) {
      prop_down_[i][b] = blob_needs[bottoms_[i][b]];
      need = need || prop_down_[i][b];
    }
    if (is_source) need = false;
    if (force && !is_source) {
      need = true;
      for (size_t b = 0; b < bottoms_[i].size(); ++b)
        prop_down_[i][b] = blob_needs.count(bottoms_[i][b])
                               ? blob_needs[bottoms_[i][b]]
                               : false;
    }
    layer_need_bwd_[i] = need;
    for (auto* t : tops_[i])
      blob_needs[t] = need || (force && !is_source &&
                               layers_[i]->type() != "SoftmaxWithLoss");
  }

  // learnable params in backward-completion order (reverse layer order)
  for (int i = (int)layers_.size() - 1; i >= 0; --i) {
    auto& lb = layers_[i]->blobs();
    for (size_t j = 0; j < lb.size(); ++j) {
      if (layers_[i]->skip_apply_update((int)j)) continue;
      LParam p;
      p.blob = lb[j].get();
      p.layer = layers_[i].get();
      p.blob_idx = (int)j;
      p.lr_mult = layers_[i]->lr_mult((int)j);
      p.decay_mult = layers_[i]->decay_mult((int)j);
      p.count = p.blob->count();
      p.offset = arena_count_;
      arena_count_ += (long)padded(p.count);
      params_.push_back(p);
    }
  }
  // GPU-mode forward fusion: an in-place slope-0 ReLU directly after a
  // BatchNorm or Eltwise(SUM) is applied in the producer's epilogue; the
  // ReLU layer keeps its backward (it reads its own output, which the
  // fused producer already wrote post-activation)
  if (Engine::get().mode == Mode::GPU) {
    for (size_t i = 1; i < layers_.size(); ++i) {
      auto* relu = dynamic_cast<ReLULayer*>(layers_[i].get());
      if (!relu) continue;
      if (bottoms_[i].size() != 1 || tops_[i].size() != 1) continue;
      if (bottoms_[i][0] != tops_[i][0]) continue;  // must be in-place
      auto rp = layers_[i]->param()->sub("relu_param");
      if (rp && rp->num("negative_slope", 0) != 0) continue;
      // producer = previous layer producing this blob
      if (tops_[i - 1].size() != 1 || tops_[i - 1][0] != bottoms_[i][0])
        continue;
      if (auto* bn = dynamic_cast<BatchNormLayer*>(layers_[i - 1].get())) {
        bn->fuse_relu_ = true;
        relu->fused_away_ = true;
        relu->bwd_fused_ = true;  // BN backward masks dy by the activation
      } else if (auto* cv =
                     dynamic_cast<ConvolutionLayer*>(layers_[i - 1].get())) {
        cv->fuse_relu_ = true;  // GemmEpi.relu (AlexNet/GoogLeNet pattern)
        relu->fused_away_ = true;
      } else if (auto* el =
                     dynamic_cast<EltwiseLayer*>(layers_[i - 1].get())) {
        bool ones = el->op_ == "SUM";
        for (float c : el->coeffs_) ones = ones && c == 1.f;
        if (ones && bottoms_[i - 1].size() == 2) {
          el->fuse_relu_ = true;
          relu->fused_away_ = true;
        }
      }
    }
    // residual-add fusion (TRAIN only; the TEST BN path has no epilogue):
    // Eltwise(SUM, coeffs 1) whose first-hand operand is produced by the
    // immediately preceding BatchNorm, and that operand has no other
    // consumer — the BN's norm epilogue writes bn(x)+other straight into
    // the sum's top (the ResNet block pattern; kills one full
    // read-add-write pass over the activation per block)
    if (phase_ == Phase::TRAIN) {
      for (size_t i = 1; i < layers_.size(); ++i) {
        auto* el = dynamic_cast<EltwiseLayer*>(layers_[i].get());
        if (!el || el->fused_away_) continue;
        if (el->op_ != "SUM" || bottoms_[i].size() != 2) continue;
        bool ones = true;
        for (float c : el->coeffs_) ones = ones && c == 1.f;
        if (!ones) continue;
        auto* bn = dynamic_cast<BatchNormLayer*>(layers_[i - 1].get());
        if (!bn || bn->fuse_relu_ || bn->fuse_add_out_) continue;
        if (tops_[i - 1].size() != 1) continue;
        Blob* bnout = tops_[i - 1][0];
        if (bnout == bottoms_[i - 1][0]) continue;  // in-place BN: keep
        int idx = bnout == bottoms_[i][0] ? 0
                  : bnout == bottoms_[i][1] ? 1
                                            : -1;
        if (idx < 0) continue;
        int consumers = 0;
        for (size_t k = 0; k < layers_.size(); ++k)
          for (auto* b : bottoms_[k]) consumers += b == bnout;
        if (consumers != 1) continue;
        bn->fuse_add_other_ = bottoms_[i][1 - idx];
        bn->fuse_add_out_ = tops_[i][0];
        bn->fuse_add_relu_ = el->fuse_relu_;
        el->fused_away_ = true;
        // absorb a following non-in-place slope-0 ReLU too: the epilogue
        // writes relu(bn(x)+other) straight into the ReLU's top and the
        // sum is never materialized (its only consumer is the ReLU, whose
        // backward masks by the top's sign instead)
        if (i + 1 < layers_.size()) {
          auto* relu = dynamic_cast<ReLULayer*>(layers_[i + 1].get());
          if (relu && !relu->fused_away_ && bottoms_[i + 1].size() == 1 &&
              tops_[i + 1].size() == 1 &&
              bottoms_[i + 1][0] == tops_[i][0] &&
              bottoms_[i + 1][0] != tops_[i + 1][0]) {
            auto rp = relu->param()->sub("relu_param");
            const bool slope0 = !rp || rp->num("negative_slope", 0) == 0;
            Blob* sum = tops_[i][0];
            int sum_consumers = 0;
            for (size_t k = 0; k < layers_.size(); ++k)
              for (auto* b : bottoms_[k]) sum_consumers += b == sum;
            if (slope0 && sum_consumers == 1) {
              bn->fuse_add_out_ = tops_[i + 1][0];
              bn->fuse_add_relu_ = true;
              relu->fused_away_ = true;
              relu->bwd_from_top_ = true;
            }
          }
        }
      }
    }
    setup_arena();
  }
}

void Net::setup_arena() {
  if (params_.empty()) return;
  Engine& E = Engine::get();
  diff_arena_ =
      (float*)E.dalloc.alloc(sizeof(float) * (size_t)arena_count_);
  HIP_CHECK(hipMemsetAsync(diff_arena_, 0,
                           sizeof(float) * (size_t)arena_count_, E.stream));
  for (auto& p : params_)
    p.blob->diff_mem().set_gpu_view(diff_arena_ + p.offset);
  layer_events_.resize(layers_.size(), nullptr);
  for (auto& ev : layer_events_)
    HIP_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
}

void Net::Forward() {
  for (size_t i = 0; i < layers_.size(); ++i)
    {
      const std::string ctx_s = "layer '" + layers_[i]->name() + "' (" +
                                layers_[i]->type() + ") Forward";
      error_context() = ctx_s.c_str();
      layers_[i]->Forward(bottoms_[i], tops_[i]);
      error_context() = nullptr;
    }
}

void Net::Backward(ReduceHook* hook) {
  Engine& E = Engine::get();
  // param ids grouped by layer, in backward order == ascending arena offset
  size_t pidx = 0;
  hipEvent_t last_ev = nullptr;
  for (int i = (int)layers_.size() - 1; i >= 0; --i) {
    if (layer_need_bwd_[i])
      {
        const std::string ctx_s = "layer '" + layers_[i]->name() + "' (" +
                                  layers_[i]->type() + ") Backward";
        error_context() = ctx_s.c_str();
        layers_[i]->Backward(tops_[i], prop_down_[i], bottoms_[i]);
        error_context() = nullptr;
      }
    if (hook) {
      // emit this layer's params (consecutive in params_, ascending offset)
      const size_t start = pidx;
      while (pidx < params_.size() &&
             params_[pidx].layer == layers_[i].get())
        ++pidx;
      if (pidx > start) {
        hipEvent_t ev = nullptr;
        if (E.mode == Mode::GPU) {
          ev = layer_events_[i];
          HIP_CHECK(hipEventRecord(ev, E.stream));
          last_ev = ev;
        }
        for (size_t k = start; k < pidx; ++k) hook->param_ready((int)k, ev);
      }
    }
  }
  if (hook) {
    hipEvent_t ev = last_ev;
    if (E.mode == Mode::GPU && !ev) {
      ev = layer_events_[0];
      HIP_CHECK(hipEventRecord(ev, E.stream));
    }
    hook->iteration_end(ev);
  }
}

float Net::loss() {
  Engine::get().sync();
  float total = 0.f;
  for (size_t i = 0; i < layers_.size(); ++i)
    for (size_t t = 0; t < tops_[i].size(); ++t) {
      const float w = layers_[i]->loss((int)t);
      if (w != 0.f && layers_[i]->type() != "Accuracy")
        total += w * tops_[i][t]->cpu_data()[0];
    }
  return total;
}

std::vector<std::pair<std::string, float>> Net::scores() {
  Engine::get().sync();
  std::vector<std::pair<std::string, float>> out;
  for (size_t i = 0; i < layers_.size(); ++i)
    for (size_t t = 0; t < tops_[i].size(); ++t) {
      const bool acc = layers_[i]->type() == "Accuracy";
      if (layers_[i]->loss((int)t) != 0.f || acc)
        out.emplace_back(layers_[i]->name(), tops_[i][t]->cpu_data()[0]);
    }
  return out;
}

std::vector<std::string> Net::blob_names() const {
  std::vector<std::string> out;
  for (auto& kv : blob_map_) out.push_back(kv.first);
  return out;
}

void Net::ShareTrainedLayersWith(Net& other) {
  std::map<std::string, Layer*> by_name;
  for (auto& l : other.layers_) by_name[l->name()] = l.get();
  for (auto& l : layers_) {
    auto it = by_name.find(l->name());
    if (it == by_name.end()) continue;
    auto& src = it->second->blobs();
    auto& dst = l->blobs();
    for (size_t i = 0; i < dst.size() && i < src.size(); ++i)
      dst[i]->ShareData(*src[i]);
  }
}

}  // namespace camd

// ------------------------------------------------- snapshot interop
// .caffemodel = binary NetParameter carrying each parametered layer's blobs
// (reference Net::ToProto / CopyTrainedLayersFrom, net.cpp:1055-1248).
#include <filesystem>
#include <fstream>

#include "proto_wire.hpp"

namespace camd {

void Net::SaveWeights(const std::string& path) {
  Engine::get().sync();
  wire::Writer net_w;
  net_w.str(1, name_);
  for (size_t i = 0; i < layers_.size(); ++i) {
    auto& lb = layers_[i]->blobs();
    if (lb.empty()) continue;
    wire::Writer lw;
    lw.str(1, layers_[i]->name());
    lw.str(2, layers_[i]->type());
    for (auto& b : lb) {
      wire::Writer bw;
      bw.packed_floats(5, b->cpu_data(), b->count());
      wire::Writer sw;
      std::vector<int64_t> dims(b->shape().begin(), b->shape().end());
      sw.packed_i64(1, dims);
      bw.submsg(7, sw.out);
      lw.submsg(7, bw.out);
    }
    net_w.submsg(100, lw.out);
  }
  const auto dir = std::filesystem::path(path).parent_path();
  if (!dir.empty()) std::filesystem::create_directories(dir);
  std::ofstream f(path, std::ios::binary);
  CHECK_(f.good()) << "cannot write " << path;
  f.write(net_w.out.data(), (long)net_w.out.size());
}

void Net::LoadWeights(const std::string& path) {
  std::ifstream f(path, std::ios::binary);
  CHECK_(f.good()) << "cannot read " << path;
  std::string buf((std::istreambuf_iterator<char>(f)),
                  std::istreambuf_iterator<char>());
  std::map<std::string, Layer*> by_name;
  for (auto& l : layers_) by_name[l->name()] = l.get();
  wire::Reader r(buf.data(), buf.size());
  wire::Field fld;
  int loaded = 0;
  while (r.next(&fld)) {
    if (fld.num != 100 || fld.wt != 2) continue;  // LayerParameter
    wire::Reader lr(fld.data, fld.len);
    wire::Field lf;
    std::string lname;
    std::vector<wire::BlobData> blobs;
    while (lr.next(&lf)) {
      if (lf.num == 1 && lf.wt == 2)
        lname.assign(lf.data, lf.len);
      else if (lf.num == 7 && lf.wt == 2)
        blobs.push_back(wire::parse_blob(lf.data, lf.len));
    }
    auto it = by_name.find(lname);
    if (it == by_name.end()) continue;  // reference skips unknown layers
    auto& lb = it->second->blobs();
    for (size_t j = 0; j < blobs.size() && j < lb.size(); ++j) {
      CHECK_EQ_((long)blobs[j].data.size(), lb[j]->count())
          << "blob size mismatch in " << lname << " blob " << j;
      memcpy(lb[j]->mutable_cpu_data(), blobs[j].data.data(),
             blobs[j].data.size() * sizeof(float));
      ++loaded;
    }
  }
  CHECK_GT_(loaded, 0) << "no matching layer blobs in " << path;
}

void Net::time_layers(int iters) {
  Engine& E = Engine::get();
  const size_t L = layers_.size();
  std::vector<double> fwd_ms(L, 0), bwd_ms(L, 0);
  auto tick = [&]() {
    E.sync();
    return std::chrono::steady_clock::now();
  };
  Forward();  // warmup + shapes
  Backward(nullptr);
  for (int it = 0; it < iters; ++it) {
    for (size_t i = 0; i < L; ++i) {
      auto t0 = tick();
      {
      const std::string ctx_s = "layer '" + layers_[i]->name() + "' (" +
                                layers_[i]->type() + ") Forward";
      error_context() = ctx_s.c_str();
      layers_[i]->Forward(bottoms_[i], tops_[i]);
      error_context() = nullptr;
    }
      auto t1 = tick();
      fwd_ms[i] += std::chrono::duration<double, std::milli>(t1 - t0).count();
    }
    for (size_t i = L; i-- > 0;) {
      if (!layer_need_bwd_[i]) continue;
      auto t0 = tick();
      {
        const std::string ctx_s = "layer '" + layers_[i]->name() + "' (" +
                                  layers_[i]->type() + ") Backward";
        error_context() = ctx_s.c_str();
        layers_[i]->Backward(tops_[i], prop_down_[i], bottoms_[i]);
        error_context() = nullptr;
      }
      auto t1 = tick();
      bwd_ms[i] += std::chrono::duration<double, std::milli>(t1 - t0).count();
    }
  }
  double ftot = 0, btot = 0;
  fprintf(stderr, "%-28s %12s %12s\n", "layer", "forward(ms)",
          "backward(ms)");
  for (size_t i = 0; i < L; ++i) {
    fprintf(stderr, "%-28s %12.3f %12.3f\n", layers_[i]->name().c_str(),
            fwd_ms[i] / iters, bwd_ms[i] / iters);
    ftot += fwd_ms[i] / iters;
    btot += bwd_ms[i] / iters;
  }
  fprintf(stderr, "%-28s %12.3f %12.3f  (total %.3f ms/iter)\n", "TOTAL",
          ftot, btot, ftot + btot);
}

}  // namespace camd
