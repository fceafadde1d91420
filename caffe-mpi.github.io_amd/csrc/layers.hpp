// layers.hpp — the drop-in boundary: Layer base with
// SetUp/Reshape/Forward_{cpu,gpu}/Backward_{cpu,gpu} over vector<Blob*> and
// a string-keyed registry, mirroring the reference's
// include/caffe/layer.hpp:43-613 + layer_factory.hpp (REGISTER_LAYER_CLASS).
// Engine is CAFFE-only (our HIP kernels) — no cuDNN-style variants.
#pragma once

#include <functional>

#include "core.hpp"
#include "math.hpp"
#include "proto.hpp"

namespace camd {

class Layer {
 public:
  explicit Layer(const PMsgPtr& param);
  virtual ~Layer() = default;

  void SetUp(const std::vector<Blob*>& bottom, const std::vector<Blob*>& top) {
    check_blob_counts(bottom, top);
    LayerSetUp(bottom, top);
    Reshape(bottom, top);
  }
  // blob-count contract (reference layer.hpp CheckBlobCounts /
  // ExactNumBottomBlobs analog): enforced before any blob is touched, so
  // a miswired prototxt (e.g. a typo'd `bottom:` key leaving a layer with
  // no input) fails with a message naming the layer instead of an
  // out-of-range access
  void check_blob_counts(const std::vector<Blob*>& bottom,
                         const std::vector<Blob*>& top) const {
    CHECK_GE_((int)bottom.size(), min_bottom_blobs())
        << type() << " layer '" << name() << "': needs at least "
        << min_bottom_blobs() << " bottom blob(s), got " << bottom.size();
    CHECK_LE_((int)bottom.size(), max_bottom_blobs())
        << type() << " layer '" << name() << "': takes at most "
        << max_bottom_blobs() << " bottom blob(s), got " << bottom.size();
    CHECK_GE_((int)top.size(), min_top_blobs())
        << type() << " layer '" << name() << "': needs at least "
        << min_top_blobs() << " top blob(s), got " << top.size();
    CHECK_LE_((int)top.size(), max_top_blobs())
        << type() << " layer '" << name() << "': takes at most "
        << max_top_blobs() << " top blob(s), got " << top.size();
  }
  virtual void LayerSetUp(const std::vector<Blob*>&,
                          const std::vector<Blob*>&) {}
  virtual void Reshape(const std::vector<Blob*>& bottom,
                       const std::vector<Blob*>& top) = 0;

  // wrappers (reference layer.hpp:555-613): dispatch on Engine mode
  void Forward(const std::vector<Blob*>& bottom,
               const std::vector<Blob*>& top);
  void Backward(const std::vector<Blob*>& top,
                const std::vector<bool>& prop_down,
                const std::vector<Blob*>& bottom);

  virtual void Forward_cpu(const std::vector<Blob*>& bottom,
                           const std::vector<Blob*>& top) = 0;
  virtual void Forward_gpu(const std::vector<Blob*>& bottom,
                           const std::vector<Blob*>& top) {
    CAMD_FATAL << type() << ": no GPU implementation (HIP path is mandatory "
                            "in GPU mode; the engine never falls back)";
  }
  virtual void Backward_cpu(const std::vector<Blob*>& top,
                            const std::vector<bool>& prop_down,
                            const std::vector<Blob*>& bottom) = 0;
  virtual void Backward_gpu(const std::vector<Blob*>&,
                            const std::vector<bool>&,
                            const std::vector<Blob*>&) {
    CAMD_FATAL << type() << ": no GPU backward implementation";
  }

  const std::string& name() const { return name_; }
  const std::string& type() const { return type_; }
  Phase phase() const { return phase_; }
  void set_phase(Phase p) { phase_ = p; }
  const PMsgPtr& param() const { return param_; }

  std::vector<std::shared_ptr<Blob>>& blobs() { return blobs_; }
  // per-param lr/decay multipliers (param { lr_mult decay_mult } specs)
  float lr_mult(int i) const;
  float decay_mult(int i) const;
  // BN statistics blobs are excluded from the optimizer/collective, matching
  // the cuDNN BN behaviour the survey prescribes
  virtual bool skip_apply_update(int /*blob_id*/) const { return false; }

  virtual bool auto_top_blobs() const { return false; }
  virtual int min_top_blobs() const { return 1; }
  virtual int max_top_blobs() const { return 1; }
  virtual int min_bottom_blobs() const { return 1; }
  virtual int max_bottom_blobs() const { return 1; }

  // loss weight per top (loss layers default 1 for top 0)
  float loss(int i) const { return i < (int)loss_.size() ? loss_[i] : 0.f; }
  void set_loss(int i, float v) {
    if ((int)loss_.size() <= i) loss_.resize(i + 1, 0.f);
    loss_[i] = v;
  }
  virtual float default_loss_weight() const { return 0.f; }

 protected:
  PMsgPtr param_;
  std::string name_, type_;
  Phase phase_ = Phase::TRAIN;
  std::vector<std::shared_ptr<Blob>> blobs_;
  std::vector<float> loss_;
};

// registry
using LayerFactory = std::function<std::shared_ptr<Layer>(const PMsgPtr&)>;
void register_layer(const std::string& type, LayerFactory f);
std::shared_ptr<Layer> create_layer(const PMsgPtr& param);

#define REGISTER_LAYER_IMPL(type, cls, uniq)                           \
  namespace {                                                          \
  struct Reg_##uniq {                                                  \
    Reg_##uniq() {                                                     \
      ::camd::register_layer(type, [](const PMsgPtr& p) {              \
        return std::shared_ptr<::camd::Layer>(new cls(p));             \
      });                                                              \
    }                                                                  \
  } reg_##uniq;                                                        \
  }
#define REGISTER_LAYER_X(type, cls, line) REGISTER_LAYER_IMPL(type, cls, line)
#define REGISTER_LAYER(type, cls) REGISTER_LAYER_X(type, cls, __LINE__)

// filler (reference include/caffe/filler.hpp: constant/uniform/gaussian/
// xavier/msra; CPU-side mt19937, seed = engine.seed + rank)
void fill_blob(Blob& b, const PMsgPtr& filler, std::mt19937_64& rng);

// shared grow-only device/host workspaces (conv col buffers etc.), the
// reference's GPUMemory::Workspace analog (util/gpu_memory.hpp:76-127)
struct Workspace {
  void* get(int slot, size_t bytes);  // device (GPU mode) / host (CPU mode)
  static Workspace& get_global();
  ~Workspace();

 private:
  struct Buf {
    void* p = nullptr;
    size_t bytes = 0;
    bool device = false;
  };
  std::map<int, Buf> bufs_;
  std::mutex mu_;
};

// ------------------------------------------------------------- layer classes
struct LmdbFeed;  // data_lmdb.cpp: reader + prefetch + transform state

class DataLayer : public Layer {  // synthetic or LMDB source (§8a a12,
                                  // §8f.1)
 public:
  using Layer::Layer;
  void LayerSetUp(const std::vector<Blob*>&,
                  const std::vector<Blob*>&) override;
  void Reshape(const std::vector<Blob*>&, const std::vector<Blob*>&) override;
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override {}
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override {}
  int min_top_blobs() const override { return 2; }
  int max_top_blobs() const override { return 2; }
  int min_bottom_blobs() const override { return 0; }
  int max_bottom_blobs() const override { return 0; }

  int batch_ = 0, C_ = 3, H_ = 224, W_ = 224;
  uint64_t iter_ = 0;

  // LMDB mode (data_param.source exists on disk — else synthetic):
  // from-scratch reader + prefetch worker + GPU uint8 transform
  void setup_lmdb(const std::string& source);
  void forward_lmdb_cpu(const std::vector<Blob*>& top);
  void forward_lmdb_gpu(const std::vector<Blob*>& top);
  std::shared_ptr<LmdbFeed> feed_;
  int u8_slot_ = -1, geo_slot_ = -1, mean_slot_ = -1;
  bool mean_uploaded_ = false;
};

class ConvolutionLayer : public Layer {
 public:
  using Layer::Layer;
  void LayerSetUp(const std::vector<Blob*>&,
                  const std::vector<Blob*>&) override;
  void Reshape(const std::vector<Blob*>&, const std::vector<Blob*>&) override;
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  // strided 1x1 dgrad without a col2im pass (zero + strided-scatter GEMM)
  void scatter_dgrad(const std::vector<Blob*>& top,
                     const std::vector<Blob*>& bottom);

  int Cout_ = 0, kh_ = 0, kw_ = 0, sh_ = 1, sw_ = 1, ph_ = 0, pw_ = 0,
      dh_ = 1, dw_ = 1, group_ = 1;
  bool bias_ = true;
  int N_ = 0, C_ = 0, H_ = 0, W_ = 0, OH_ = 0, OW_ = 0;
  long S_ = 0, Spad_ = 0;  // spatial count and 64-padded count
  int col_slot_ = -1;      // per-layer cached col buffer (fwd fills,
                           // bwd reuses — no im2col recompute)
  bool fuse_relu_ = false;  // in-place ReLU folded into the GEMM epilogue
};

class InnerProductLayer : public Layer {
 public:
  using Layer::Layer;
  void LayerSetUp(const std::vector<Blob*>&,
                  const std::vector<Blob*>&) override;
  void Reshape(const std::vector<Blob*>&, const std::vector<Blob*>&) override;
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  int Nout_ = 0;
  long M_ = 0, K_ = 0;
  bool bias_ = true;
  Blob ones_;  // for CPU bias path
};

class PoolingLayer : public Layer {
 public:
  using Layer::Layer;
  void LayerSetUp(const std::vector<Blob*>&,
                  const std::vector<Blob*>&) override;
  void Reshape(const std::vector<Blob*>&, const std::vector<Blob*>&) override;
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  bool max_ = true;
  bool global_ = false;
  int kh_ = 0, kw_ = 0, sh_ = 1, sw_ = 1, ph_ = 0, pw_ = 0;
  int N_ = 0, C_ = 0, H_ = 0, W_ = 0, OH_ = 0, OW_ = 0;
  Blob mask_;  // int mask stored as float blob's memory (int-sized)
};

class BatchNormLayer : public Layer {
 public:
  using Layer::Layer;
  void LayerSetUp(const std::vector<Blob*>&,
                  const std::vector<Blob*>&) override;
  void Reshape(const std::vector<Blob*>&, const std::vector<Blob*>&) override;
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  bool skip_apply_update(int i) const override { return i < 3; }

  float maf_ = 0.999f, eps_ = 1e-5f;
  bool scale_bias_ = false, use_global_ = false;
  bool fuse_relu_ = false;  // GPU graph fusion: BN+ReLU forward in one pass
  // GPU graph fusion: a following Eltwise(SUM) absorbed into the norm
  // epilogue — forward writes fuse_add_out_ = bn(x) + fuse_add_other_
  // (bit-identical to the separate add; backward is untouched, the
  // eltwise aliases the sum's diff onto this layer's top)
  Blob* fuse_add_other_ = nullptr;
  Blob* fuse_add_out_ = nullptr;
  bool fuse_add_relu_ = false;
  int C_ = 0;
  long iter_ = 0;
  Blob mean_, var_, inv_std_, m_dy_, m_dyxn_, partials_;
  // in-place BN (top==bottom, the standard BVLC prototxt idiom): backward
  // recomputes x̂ from the ORIGINAL input, which the in-place forward
  // overwrites — Forward saves a copy into saved_x_ for that case only
  Blob saved_x_;
  // the input tensor backward must normalize against (bottom data, or
  // saved_x_ when in-place) — set by Forward each iteration
  const float* bwd_x(const std::vector<Blob*>& bottom,
                     const std::vector<Blob*>& top, bool gpu) {
    if (top[0] != bottom[0]) return gpu ? bottom[0]->gpu_data()
                                        : bottom[0]->cpu_data();
    return gpu ? saved_x_.gpu_data() : saved_x_.cpu_data();
  }
};

// Scale / Bias (reference layers/scale_layer.cpp, bias_layer.cpp): the
// BVLC-style companion of a stat-only BatchNorm — y = x * scale[c]
// (+ bias[c]).  Channel axis only (axis 1, num_axes 1), single bottom
// with learnable params — the pattern every public BVLC-format ResNet
// prototxt uses; the fork's own models use batch_norm_param.scale_bias
// instead (csrc BatchNormLayer).
class ScaleLayer : public Layer {
 public:
  using Layer::Layer;
  void LayerSetUp(const std::vector<Blob*>&,
                  const std::vector<Blob*>&) override;
  void Reshape(const std::vector<Blob*>& b,
               const std::vector<Blob*>& t) override {
    if (b[0] != t[0]) t[0]->ReshapeLike(*b[0]);
  }
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  bool bias_ = false;  // scale_param.bias_term
  int C_ = 0;
  Blob partials_;  // per-channel-slice double2 {sum dy*x, sum dy}
  Blob temp_;      // in-place input copy for backward (scale_layer.cpp)
  Blob zo_;        // GPU consts+scratch: [zeros C][ones C][scratch 2C]
};

// Bias (reference layers/bias_layer.cpp): y = x + bias[c], channel axis,
// single-bottom learnable form — Scale's standalone sibling
class BiasLayer : public Layer {
 public:
  using Layer::Layer;
  void LayerSetUp(const std::vector<Blob*>&,
                  const std::vector<Blob*>&) override;
  void Reshape(const std::vector<Blob*>& b,
               const std::vector<Blob*>& t) override {
    if (b[0] != t[0]) t[0]->ReshapeLike(*b[0]);
  }
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  int C_ = 0;
  Blob partials_, zo_;  // per-channel reduction workspace (GPU)
};

class ReLULayer : public Layer {
 public:
  using Layer::Layer;
  bool fused_away_ = false;  // producer applies the ReLU in its epilogue
  bool bwd_fused_ = false;   // producer also absorbs the backward mask
  // non-in-place fused ReLU whose bottom (the pre-activation) is never
  // materialized: backward masks by the TOP's sign — identical for
  // slope 0 (top > 0 <=> bottom > 0)
  bool bwd_from_top_ = false;
  void Reshape(const std::vector<Blob*>& b,
               const std::vector<Blob*>& t) override {
    if (b[0] != t[0]) t[0]->ReshapeLike(*b[0]);
  }
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
};

class EltwiseLayer : public Layer {
 public:
  using Layer::Layer;
  int min_bottom_blobs() const override { return 2; }
  int max_bottom_blobs() const override { return 4096; }
  void LayerSetUp(const std::vector<Blob*>&,
                  const std::vector<Blob*>&) override;
  void Reshape(const std::vector<Blob*>& b,
               const std::vector<Blob*>& t) override {
    // eltwise_layer.cpp Reshape: all bottoms must match exactly
    for (size_t i = 1; i < b.size(); ++i)
      CHECK_(b[i]->shape() == b[0]->shape())
          << "Eltwise bottoms must have identical shapes";
    t[0]->ReshapeLike(*b[0]);
  }
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  std::vector<float> coeffs_;
  std::string op_ = "SUM";
  bool fuse_relu_ = false;  // GPU graph fusion: SUM+ReLU forward in one pass
  bool fused_away_ = false;  // producing BN writes bn(x)+other directly
};

class LRNLayer : public Layer {
 public:
  using Layer::Layer;
  void LayerSetUp(const std::vector<Blob*>&,
                  const std::vector<Blob*>&) override;
  void Reshape(const std::vector<Blob*>& b,
               const std::vector<Blob*>& t) override {
    t[0]->ReshapeLike(*b[0]);
    scale_.ReshapeLike(*b[0]);
  }
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  int size_ = 5;
  float alpha_ = 1.f, beta_ = 0.75f, k_ = 1.f;
  Blob scale_;
};

class DropoutLayer : public Layer {
 public:
  using Layer::Layer;
  void LayerSetUp(const std::vector<Blob*>&,
                  const std::vector<Blob*>&) override;
  void Reshape(const std::vector<Blob*>& b,
               const std::vector<Blob*>& t) override {
    if (b[0] != t[0]) t[0]->ReshapeLike(*b[0]);
    mask_.Reshape({(int)((b[0]->count() + 3) / 4)});  // uint8 mask storage
  }
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  float ratio_ = 0.5f, scale_ = 2.f;
  uint64_t iter_ = 0;
  Blob mask_;
};

class ConcatLayer : public Layer {
 public:
  using Layer::Layer;
  int max_bottom_blobs() const override { return 4096; }
  void Reshape(const std::vector<Blob*>&, const std::vector<Blob*>&) override;
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
};

class SplitLayer : public Layer {
 public:
  using Layer::Layer;
  int max_top_blobs() const override { return 4096; }
  void Reshape(const std::vector<Blob*>&, const std::vector<Blob*>&) override;
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>& b,
                   const std::vector<Blob*>& t) override {
    Forward_cpu(b, t);  // pure blob sharing, no kernels
  }
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
};

class SoftmaxLayer : public Layer {
 public:
  using Layer::Layer;
  void Reshape(const std::vector<Blob*>& b,
               const std::vector<Blob*>& t) override;
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override {
    CAMD_FATAL << "Softmax GPU backward unused by the four models";
  }
  int outer_ = 0, C_ = 0, inner_ = 0;
};

class SoftmaxWithLossLayer : public Layer {
 public:
  using Layer::Layer;
  int min_bottom_blobs() const override { return 2; }
  int max_bottom_blobs() const override { return 2; }
  void LayerSetUp(const std::vector<Blob*>&,
                  const std::vector<Blob*>&) override;
  void Reshape(const std::vector<Blob*>&, const std::vector<Blob*>&) override;
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override;
  float default_loss_weight() const override { return 1.f; }
  int outer_ = 0, C_ = 0, inner_ = 0;
  Blob prob_;
};

class AccuracyLayer : public Layer {  // CPU-resident (reference: CPU only)
 public:
  using Layer::Layer;
  int min_bottom_blobs() const override { return 2; }
  int max_bottom_blobs() const override { return 2; }
  void Reshape(const std::vector<Blob*>&, const std::vector<Blob*>&) override;
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override;
  void Forward_gpu(const std::vector<Blob*>& b,
                   const std::vector<Blob*>& t) override {
    Forward_cpu(b, t);  // pulls data to host; test-phase only
  }
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override {}
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override {}
  int top_k_ = 1;
};

}  // namespace camd
