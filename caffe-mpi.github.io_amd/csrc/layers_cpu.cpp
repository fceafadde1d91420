// layers_cpu.cpp — Layer base, registry, fillers, workspace and the CPU
// implementations (the engine's own CPU mode for BASELINE config 1; GPU
// parity is checked against oracle/, never against this file).
#include <cmath>

#include <sys/stat.h>

#include "layers.hpp"

namespace camd {

// ------------------------------------------------------------- Layer base
Layer::Layer(const PMsgPtr& param) : param_(param) {
  name_ = param->str("name");
  type_ = param->str("type");
}

float Layer::lr_mult(int i) const {
  auto specs = param_->subs("param");
  if (i < (int)specs.size()) return (float)specs[i]->num("lr_mult", 1.0);
  return 1.f;
}
float Layer::decay_mult(int i) const {
  auto specs = param_->subs("param");
  if (i < (int)specs.size()) return (float)specs[i]->num("decay_mult", 1.0);
  return 1.f;
}

void Layer::Forward(const std::vector<Blob*>& bottom,
                    const std::vector<Blob*>& top) {
  Reshape(bottom, top);
  if (Engine::get().mode == Mode::GPU)
    Forward_gpu(bottom, top);
  else
    Forward_cpu(bottom, top);
}

void Layer::Backward(const std::vector<Blob*>& top,
                     const std::vector<bool>& prop_down,
                     const std::vector<Blob*>& bottom) {
  if (Engine::get().mode == Mode::GPU)
    Backward_gpu(top, prop_down, bottom);
  else
    Backward_cpu(top, prop_down, bottom);
}

// --------------------------------------------------------------- registry
static std::map<std::string, LayerFactory>& registry() {
  static std::map<std::string, LayerFactory> r;
  return r;
}
void register_layer(const std::string& type, LayerFactory f) {
  registry()[type] = std::move(f);
}
std::shared_ptr<Layer> create_layer(const PMsgPtr& param) {
  const std::string type = param->str("type");
  auto it = registry().find(type);
  CHECK_(it != registry().end()) << "unknown layer type: " << type;
  return it->second(param);
}

// ----------------------------------------------------------------- filler
void fill_blob(Blob& b, const PMsgPtr& filler, std::mt19937_64& rng) {
  const std::string type = filler ? filler->str("type", "constant")
                                  : "constant";
  float* p = b.mutable_cpu_data();
  const long n = b.count();
  if (type == "constant") {
    const float v = filler ? (float)filler->num("value", 0) : 0.f;
    for (long i = 0; i < n; ++i) p[i] = v;
  } else if (type == "uniform") {
    std::uniform_real_distribution<float> d(
        (float)filler->num("min", 0), (float)filler->num("max", 1));
    for (long i = 0; i < n; ++i) p[i] = d(rng);
  } else if (type == "gaussian") {
    std::normal_distribution<float> d((float)filler->num("mean", 0),
                                      (float)filler->num("std", 1));
    for (long i = 0; i < n; ++i) p[i] = d(rng);
  } else if (type == "xavier" || type == "msra") {
    // filler.hpp:278-300 (xavier) / MSRAFiller: n = fan_in default,
    // AVERAGE/FAN_OUT via variance_norm
    const long fan_in = b.count() / b.num();
    const long fan_out = b.count() / b.channels();
    double fan = fan_in;
    const std::string vn = filler->str("variance_norm", "FAN_IN");
    if (vn == "AVERAGE") fan = (fan_in + fan_out) / 2.0;
    else if (vn == "FAN_OUT") fan = fan_out;
    if (type == "xavier") {
      const float s = std::sqrt(3.0 / fan);
      std::uniform_real_distribution<float> d(-s, s);
      for (long i = 0; i < n; ++i) p[i] = d(rng);
    } else {
      std::normal_distribution<float> d(0.f, (float)std::sqrt(2.0 / fan));
      for (long i = 0; i < n; ++i) p[i] = d(rng);
    }
  } else {
    CAMD_FATAL << "unsupported filler type: " << type;
  }
}

// -------------------------------------------------------------- workspace
Workspace& Workspace::get_global() {
  static Workspace w;
  return w;
}
void* Workspace::get(int slot, size_t bytes) {
  std::lock_guard<std::mutex> g(mu_);
  Buf& b = bufs_[slot];
  const bool want_dev = Engine::get().mode == Mode::GPU;
  if (b.p && (b.bytes < bytes || b.device != want_dev)) {
    if (b.device)
      Engine::get().dalloc.release(b.p, b.bytes);
    else
      free(b.p);
    b.p = nullptr;
  }
  if (!b.p) {
    b.bytes = bytes;
    b.device = want_dev;
    b.p = want_dev ? Engine::get().dalloc.alloc(bytes) : malloc(bytes);
    CHECK_(b.p) << "workspace alloc failed (" << bytes << " B)";
  }
  return b.p;
}
Workspace::~Workspace() {
  for (auto& kv : bufs_)
    if (kv.second.p && !kv.second.device) free(kv.second.p);
}


// ---------------------------------------------------------------- Data
void DataLayer::LayerSetUp(const std::vector<Blob*>&,
                           const std::vector<Blob*>&) {
  auto dp = param_->sub("data_param");
  batch_ = dp ? (int)dp->inum("batch_size", 1) : 1;
  // LMDB source when data_param.source EXISTS on disk (reference
  // db_lmdb.cpp backend); a prototxt naming an absent dataset falls back
  // to the synthetic stream so the reference model zoo runs unmodified
  // in this dataset-less environment (round-1 drop-in contract)
  const std::string src = dp ? dp->str("source", "") : "";
  if (!src.empty()) {
    struct stat st {};
    if (stat(src.c_str(), &st) == 0) {
      setup_lmdb(src);
      return;
    }
    fprintf(stderr,
            "[caffe_amd] DataLayer '%s': source '%s' not found — using "
            "the synthetic stream\n",
            name_.c_str(), src.c_str());
  }
  auto tp = param_->sub("transform_param");
  const int crop = tp ? (int)tp->inum("crop_size", 0) : 0;
  // synthetic shape: explicit engine override, else crop_size (3 channels)
  extern int g_syn_shape[3];
  if (g_syn_shape[1] > 0) {
    C_ = g_syn_shape[0];
    H_ = g_syn_shape[1];
    W_ = g_syn_shape[2];
  } else if (crop > 0) {
    C_ = 3;
    H_ = W_ = crop;
  } else {
    CAMD_FATAL << "DataLayer '" << name_
               << "': no crop_size and no synthetic shape set "
                  "(caffe_set_synthetic_shape)";
  }
}
int g_syn_shape[3] = {0, 0, 0};

void DataLayer::Reshape(const std::vector<Blob*>&,
                        const std::vector<Blob*>& top) {
  top[0]->Reshape({batch_, C_, H_, W_});
  top[1]->Reshape({batch_});
}

void DataLayer::Forward_cpu(const std::vector<Blob*>&,
                            const std::vector<Blob*>& top) {
  if (feed_) {
    forward_lmdb_cpu(top);
    ++iter_;
    return;
  }
  Engine& E = Engine::get();
  const uint64_t key = h_splitmix64(E.seed ^ ((uint64_t)E.rank << 40) ^
                                  (E.data_iter << 8));
  float* d = top[0]->mutable_cpu_data();
  const long n = top[0]->count();
  for (long i = 0; i < n; ++i)
    d[i] = h_u01(h_splitmix64(key ^ (uint64_t)i)) * 2.f - 1.f;
  float* l = top[1]->mutable_cpu_data();
  for (long i = 0; i < batch_; ++i)
    l[i] = (float)(h_splitmix64(key ^ 0xABCDull ^ (uint64_t)i) %
                   (uint64_t)E.syn_classes);
  ++iter_;
}

// ---------------------------------------------------------------- Conv
void ConvolutionLayer::LayerSetUp(const std::vector<Blob*>& bottom,
                                  const std::vector<Blob*>&) {
  auto cp = param_->sub("convolution_param");
  CHECK_(cp) << "Convolution layer needs convolution_param";
  Cout_ = (int)cp->inum("num_output");
  CHECK_GT_(Cout_, 0) << "convolution_param.num_output must be positive";
  const int k = (int)cp->inum("kernel_size", 0);
  kh_ = k ? k : (int)cp->inum("kernel_h");
  kw_ = k ? k : (int)cp->inum("kernel_w");
  CHECK_GT_(kh_, 0);
  const int s = (int)cp->inum("stride", 1);
  sh_ = cp->has("stride_h") ? (int)cp->inum("stride_h") : s;
  sw_ = cp->has("stride_w") ? (int)cp->inum("stride_w") : s;
  const int p = (int)cp->inum("pad", 0);
  ph_ = cp->has("pad_h") ? (int)cp->inum("pad_h") : p;
  pw_ = cp->has("pad_w") ? (int)cp->inum("pad_w") : p;
  dh_ = dw_ = (int)cp->inum("dilation", 1);
  group_ = (int)cp->inum("group", 1);
  bias_ = cp->boolean("bias_term", true);
  C_ = bottom[0]->channels();
  CHECK_EQ_(C_ % group_, 0);
  CHECK_EQ_(Cout_ % group_, 0);
  if (blobs_.empty()) {
    blobs_.emplace_back(
        new Blob({Cout_, C_ / group_, kh_, kw_}));
    if (bias_) blobs_.emplace_back(new Blob({Cout_}));
    Engine& E = Engine::get();
    fill_blob(*blobs_[0], cp->sub("weight_filler"), E.cpu_rng);
    if (bias_) fill_blob(*blobs_[1], cp->sub("bias_filler"), E.cpu_rng);
  }
}

void ConvolutionLayer::Reshape(const std::vector<Blob*>& bottom,
                               const std::vector<Blob*>& top) {
  N_ = bottom[0]->num();
  C_ = bottom[0]->channels();
  H_ = bottom[0]->height();
  W_ = bottom[0]->width();
  OH_ = conv_out_dim(H_, kh_, ph_, sh_, dh_);
  OW_ = conv_out_dim(W_, kw_, pw_, sw_, dw_);
  CHECK_GT_(OH_, 0);
  CHECK_GT_(OW_, 0);
  S_ = (long)OH_ * OW_;
  // pad S to 16 (the GEMM staging chunk width): view chunks stay inside one
  // image and padding waste is <= 15 columns per image
  Spad_ = (S_ + 15) / 16 * 16;
  top[0]->Reshape({N_, Cout_, OH_, OW_});
}

void ConvolutionLayer::Forward_cpu(const std::vector<Blob*>& bottom,
                                   const std::vector<Blob*>& top) {
  const int K = C_ / group_ * kh_ * kw_;
  float* col = (float*)Workspace::get_global().get(
      0, sizeof(float) * (size_t)C_ * kh_ * kw_ * S_);
  const float* x = bottom[0]->cpu_data();
  const float* w = blobs_[0]->cpu_data();
  float* y = top[0]->mutable_cpu_data();
  for (int n = 0; n < N_; ++n) {
    const float* xn = x + (long)n * C_ * H_ * W_;
    const float* colp = col;
    if (kh_ == 1 && kw_ == 1 && sh_ == 1 && sw_ == 1 && !ph_ && !pw_) {
      colp = xn;  // 1x1/s1 skips im2col (base_conv_layer.hpp:99-103)
    } else {
      cpu::im2col(xn, C_, H_, W_, kh_, kw_, ph_, pw_, sh_, sw_, dh_, dw_,
                  col);
    }
    for (int g = 0; g < group_; ++g)
      cpu::gemm(false, false, Cout_ / group_, S_, K, 1.f,
                w + (long)g * (Cout_ / group_) * K, colp + (long)g * K * S_,
                0.f, y + ((long)n * Cout_ + (long)g * (Cout_ / group_)) * S_);
    if (bias_) {
      const float* b = blobs_[1]->cpu_data();
      float* yn = y + (long)n * Cout_ * S_;
      for (int co = 0; co < Cout_; ++co)
        for (long s = 0; s < S_; ++s) yn[co * S_ + s] += b[co];
    }
  }
}

void ConvolutionLayer::Backward_cpu(const std::vector<Blob*>& top,
                                    const std::vector<bool>& prop_down,
                                    const std::vector<Blob*>& bottom) {
  const int K = C_ / group_ * kh_ * kw_;
  const size_t colbytes = sizeof(float) * (size_t)C_ * kh_ * kw_ * S_;
  float* col = (float*)Workspace::get_global().get(0, colbytes);
  float* dcol = (float*)Workspace::get_global().get(1, colbytes);
  const float* x = bottom[0]->cpu_data();
  const float* w = blobs_[0]->cpu_data();
  const float* dy = top[0]->cpu_diff();
  float* dw = blobs_[0]->mutable_cpu_diff();
  float* dx = prop_down[0] ? bottom[0]->mutable_cpu_diff() : nullptr;
  memset(dw, 0, sizeof(float) * blobs_[0]->count());
  if (bias_) {
    float* db = blobs_[1]->mutable_cpu_diff();
    memset(db, 0, sizeof(float) * Cout_);
    for (int n = 0; n < N_; ++n) {
      const float* dyn = dy + (long)n * Cout_ * S_;
      for (int co = 0; co < Cout_; ++co) {
        float acc = 0;
        for (long s = 0; s < S_; ++s) acc += dyn[co * S_ + s];
        db[co] += acc;
      }
    }
  }
  const bool is1x1 =
      kh_ == 1 && kw_ == 1 && sh_ == 1 && sw_ == 1 && !ph_ && !pw_;
  for (int n = 0; n < N_; ++n) {
    const float* xn = x + (long)n * C_ * H_ * W_;
    const float* dyn = dy + (long)n * Cout_ * S_;
    const float* colp = is1x1 ? xn : col;
    if (!is1x1)
      cpu::im2col(xn, C_, H_, W_, kh_, kw_, ph_, pw_, sh_, sw_, dh_, dw_,
                  col);
    for (int g = 0; g < group_; ++g) {
      cpu::gemm(false, true, Cout_ / group_, K, S_, 1.f,
                dyn + (long)g * (Cout_ / group_) * S_, colp + (long)g * K * S_,
                1.f, dw + (long)g * (Cout_ / group_) * K);
      if (dx)
        cpu::gemm(true, false, K, S_, Cout_ / group_, 1.f,
                  w + (long)g * (Cout_ / group_) * K,
                  dyn + (long)g * (Cout_ / group_) * S_, 0.f,
                  (is1x1 ? dx + (long)n * C_ * H_ * W_ : dcol) +
                      (long)g * K * S_);
    }
    if (dx && !is1x1)
      cpu::col2im(dcol, C_, H_, W_, kh_, kw_, ph_, pw_, sh_, sw_, dh_, dw_,
                  dx + (long)n * C_ * H_ * W_);
  }
}

// ---------------------------------------------------------------- IP
void InnerProductLayer::LayerSetUp(const std::vector<Blob*>& bottom,
                                   const std::vector<Blob*>&) {
  auto ip = param_->sub("inner_product_param");
  CHECK_(ip);
  Nout_ = (int)ip->inum("num_output");
  CHECK_GT_(Nout_, 0) << "inner_product_param.num_output must be positive";
  bias_ = ip->boolean("bias_term", true);
  // flatten-from-axis-1, W stored [Nout][K] — any other axis or the
  // transposed-weight layout must fail loudly, not silently mis-multiply
  CHECK_EQ_(ip->inum("axis", 1), 1)
      << "only inner_product_param.axis 1 is implemented";
  CHECK_(!ip->boolean("transpose", false))
      << "inner_product_param.transpose is not implemented";
  K_ = bottom[0]->count(1);
  if (blobs_.empty()) {
    blobs_.emplace_back(new Blob({Nout_, (int)K_}));
    if (bias_) blobs_.emplace_back(new Blob({Nout_}));
    Engine& E = Engine::get();
    fill_blob(*blobs_[0], ip->sub("weight_filler"), E.cpu_rng);
    if (bias_) fill_blob(*blobs_[1], ip->sub("bias_filler"), E.cpu_rng);
  }
}

void InnerProductLayer::Reshape(const std::vector<Blob*>& bottom,
                                const std::vector<Blob*>& top) {
  M_ = bottom[0]->num();
  CHECK_EQ_(bottom[0]->count(1), K_);
  top[0]->Reshape({(int)M_, Nout_});
}

void InnerProductLayer::Forward_cpu(const std::vector<Blob*>& bottom,
                                    const std::vector<Blob*>& top) {
  const float* x = bottom[0]->cpu_data();
  const float* w = blobs_[0]->cpu_data();
  float* y = top[0]->mutable_cpu_data();
  cpu::gemm(false, true, M_, Nout_, K_, 1.f, x, w, 0.f, y);
  if (bias_) {
    const float* b = blobs_[1]->cpu_data();
    for (long m = 0; m < M_; ++m)
      for (int n = 0; n < Nout_; ++n) y[m * Nout_ + n] += b[n];
  }
}

void InnerProductLayer::Backward_cpu(const std::vector<Blob*>& top,
                                     const std::vector<bool>& prop_down,
                                     const std::vector<Blob*>& bottom) {
  const float* dy = top[0]->cpu_diff();
  const float* x = bottom[0]->cpu_data();
  const float* w = blobs_[0]->cpu_data();
  cpu::gemm(true, false, Nout_, K_, M_, 1.f, dy, x, 0.f,
            blobs_[0]->mutable_cpu_diff());
  if (bias_) {
    float* db = blobs_[1]->mutable_cpu_diff();
    for (int n = 0; n < Nout_; ++n) {
      float acc = 0;
      for (long m = 0; m < M_; ++m) acc += dy[m * Nout_ + n];
      db[n] = acc;
    }
  }
  if (prop_down[0])
    cpu::gemm(false, false, M_, K_, Nout_, 1.f, dy, w, 0.f,
              bottom[0]->mutable_cpu_diff());
}

// ---------------------------------------------------------------- Pooling
void PoolingLayer::LayerSetUp(const std::vector<Blob*>&,
                              const std::vector<Blob*>&) {
  auto pp = param_->sub("pooling_param");
  CHECK_(pp);
  const std::string pool = pp->str("pool", "MAX");
  max_ = pool == "MAX";
  CHECK_(max_ || pool == "AVE") << "pool method " << pool;
  global_ = pp->boolean("global_pooling", false);
  if (!global_) {
    const int k = (int)pp->inum("kernel_size", 0);
    kh_ = pp->has("kernel_h") ? (int)pp->inum("kernel_h") : k;
    kw_ = pp->has("kernel_w") ? (int)pp->inum("kernel_w") : k;
    CHECK_GT_(kh_, 0);
  }
  const int s = (int)pp->inum("stride", 1);
  sh_ = pp->has("stride_h") ? (int)pp->inum("stride_h") : s;
  sw_ = pp->has("stride_w") ? (int)pp->inum("stride_w") : s;
  const int p = (int)pp->inum("pad", 0);
  ph_ = pp->has("pad_h") ? (int)pp->inum("pad_h") : p;
  pw_ = pp->has("pad_w") ? (int)pp->inum("pad_w") : p;
  if (!global_) {
    // reference pooling_layer.cpp CHECK_LT(pad, kernel): a window fully
    // inside the padding would MAX over nothing
    CHECK_LT_(ph_, kh_) << "pooling pad_h must be < kernel_h";
    CHECK_LT_(pw_, kw_) << "pooling pad_w must be < kernel_w";
  }
}

void PoolingLayer::Reshape(const std::vector<Blob*>& bottom,
                           const std::vector<Blob*>& top) {
  N_ = bottom[0]->num();
  C_ = bottom[0]->channels();
  H_ = bottom[0]->height();
  W_ = bottom[0]->width();
  if (global_) {
    kh_ = H_;
    kw_ = W_;
  }
  pool_out_dim(H_, W_, kh_, kw_, ph_, pw_, sh_, sw_, &OH_, &OW_);
  CHECK_GT_(OH_, 0) << "pooling '" << name_ << "': kernel " << kh_ << "x"
                    << kw_ << " larger than padded input " << H_ << "x"
                    << W_;
  CHECK_GT_(OW_, 0) << "pooling '" << name_ << "': kernel " << kh_ << "x"
                    << kw_ << " larger than padded input " << H_ << "x"
                    << W_;
  top[0]->Reshape({N_, C_, OH_, OW_});
  if (max_) mask_.Reshape({N_, C_, OH_, OW_});
}

void PoolingLayer::Forward_cpu(const std::vector<Blob*>& bottom,
                               const std::vector<Blob*>& top) {
  const float* x = bottom[0]->cpu_data();
  float* y = top[0]->mutable_cpu_data();
  int* mask = max_ ? (int*)mask_.mutable_cpu_data() : nullptr;
#pragma omp parallel for collapse(2) schedule(static)
  for (int n = 0; n < N_; ++n) {
    for (int c = 0; c < C_; ++c) {
      const float* xp = x + ((long)n * C_ + c) * H_ * W_;
      float* yp = y + ((long)n * C_ + c) * OH_ * OW_;
      int* mp = mask ? mask + ((long)n * C_ + c) * OH_ * OW_ : nullptr;
      for (int oh = 0; oh < OH_; ++oh)
        for (int ow = 0; ow < OW_; ++ow) {
          int hs = oh * sh_ - ph_, ws = ow * sw_ - pw_;
          if (max_) {
            int he = std::min(hs + kh_, H_), we = std::min(ws + kw_, W_);
            hs = std::max(hs, 0);
            ws = std::max(ws, 0);
            float best = -3.402823e38f;
            int bi = -1;
            for (int h = hs; h < he; ++h)
              for (int w = ws; w < we; ++w)
                if (xp[h * W_ + w] > best) {
                  best = xp[h * W_ + w];
                  bi = h * W_ + w;
                }
            yp[oh * OW_ + ow] = best;
            mp[oh * OW_ + ow] = bi;
          } else {
            int he = std::min(hs + kh_, H_ + ph_),
                we = std::min(ws + kw_, W_ + pw_);
            const int ps = (he - hs) * (we - ws);
            hs = std::max(hs, 0);
            ws = std::max(ws, 0);
            he = std::min(he, H_);
            we = std::min(we, W_);
            float acc = 0;
            for (int h = hs; h < he; ++h)
              for (int w = ws; w < we; ++w) acc += xp[h * W_ + w];
            yp[oh * OW_ + ow] = acc / ps;
          }
        }
    }
  }
}

void PoolingLayer::Backward_cpu(const std::vector<Blob*>& top,
                                const std::vector<bool>& prop_down,
                                const std::vector<Blob*>& bottom) {
  if (!prop_down[0]) return;
  const float* dy = top[0]->cpu_diff();
  float* dx = bottom[0]->mutable_cpu_diff();
  memset(dx, 0, sizeof(float) * bottom[0]->count());
  if (max_) {
    const int* mask = (const int*)mask_.cpu_data();
    for (long nc = 0; nc < (long)N_ * C_; ++nc) {
      const float* dyp = dy + nc * OH_ * OW_;
      const int* mp = mask + nc * OH_ * OW_;
      float* dxp = dx + nc * H_ * W_;
      for (long i = 0; i < (long)OH_ * OW_; ++i) dxp[mp[i]] += dyp[i];
    }
  } else {
    for (long nc = 0; nc < (long)N_ * C_; ++nc) {
      const float* dyp = dy + nc * OH_ * OW_;
      float* dxp = dx + nc * H_ * W_;
      for (int oh = 0; oh < OH_; ++oh)
        for (int ow = 0; ow < OW_; ++ow) {
          int hs = oh * sh_ - ph_, ws = ow * sw_ - pw_;
          int he = std::min(hs + kh_, H_ + ph_),
              we = std::min(ws + kw_, W_ + pw_);
          const int ps = (he - hs) * (we - ws);
          hs = std::max(hs, 0);
          ws = std::max(ws, 0);
          he = std::min(he, H_);
          we = std::min(we, W_);
          const float v = dyp[oh * OW_ + ow] / ps;
          for (int h = hs; h < he; ++h)
            for (int w = ws; w < we; ++w) dxp[h * W_ + w] += v;
        }
    }
  }
}

// ---------------------------------------------------------------- BN
void BatchNormLayer::LayerSetUp(const std::vector<Blob*>& bottom,
                                const std::vector<Blob*>&) {
  auto bp = param_->sub("batch_norm_param");
  maf_ = bp ? (float)bp->num("moving_average_fraction", 0.999) : 0.999f;
  eps_ = bp ? (float)bp->num("eps", 1e-5) : 1e-5f;
  eps_ = std::max(eps_, 1e-5f);  // batch_norm_layer.cpp:25
  scale_bias_ = bp && (bp->boolean("scale_bias", false) ||
                       bp->has("scale_filler") || bp->has("bias_filler"));
  // parsed for spec completeness but, exactly like the reference, never
  // consulted: NVCaffe-0.16 assigns use_global_stats_ (batch_norm_layer
  // .cpp:18) and then dispatches purely on phase (== TEST at :154); we
  // mirror that so a prototxt setting it behaves identically
  use_global_ = bp && bp->boolean("use_global_stats", false);
  C_ = bottom[0]->channels();
  if (blobs_.empty()) {
    blobs_.resize(scale_bias_ ? 5 : 3);
    blobs_[0].reset(new Blob({C_}));  // global mean
    blobs_[1].reset(new Blob({C_}));  // global var
    blobs_[2].reset(new Blob({1}));   // variance correction
    blobs_[0]->set_data_const(0);
    blobs_[1]->set_data_const(0);
    blobs_[2]->set_data_const(1);
    if (scale_bias_) {
      blobs_[3].reset(new Blob({C_}));
      blobs_[4].reset(new Blob({C_}));
      Engine& E = Engine::get();
      if (bp->has("scale_filler"))
        fill_blob(*blobs_[3], bp->sub("scale_filler"), E.cpu_rng);
      else
        blobs_[3]->set_data_const(1);
      if (bp->has("bias_filler"))
        fill_blob(*blobs_[4], bp->sub("bias_filler"), E.cpu_rng);
      else
        blobs_[4]->set_data_const(0);
    }
  }
}

void BatchNormLayer::Reshape(const std::vector<Blob*>& bottom,
                             const std::vector<Blob*>& top) {
  CHECK_EQ_(bottom[0]->channels(), C_);
  top[0]->ReshapeLike(*bottom[0]);
  mean_.Reshape({C_});
  var_.Reshape({C_});
  inv_std_.Reshape({C_});
  m_dy_.Reshape({C_});
  m_dyxn_.Reshape({C_});
}

void BatchNormLayer::Forward_cpu(const std::vector<Blob*>& bottom,
                                 const std::vector<Blob*>& top) {
  const int N = bottom[0]->num();
  const long S = bottom[0]->count() / ((long)N * C_);
  const float* x = bottom[0]->cpu_data();
  float* y = top[0]->mutable_cpu_data();
  if (top[0] == bottom[0] && phase_ == Phase::TRAIN) {
    // in-place BN: backward needs the original x (it recomputes x̂) —
    // save it before the normalize overwrites the blob
    saved_x_.ReshapeLike(*bottom[0]);
    memcpy(saved_x_.mutable_cpu_data(), x,
           sizeof(float) * bottom[0]->count());
  }
  const float* sc = scale_bias_ ? blobs_[3]->cpu_data() : nullptr;
  const float* bi = scale_bias_ ? blobs_[4]->cpu_data() : nullptr;
  if (phase_ == Phase::TEST) {
    const float* gm = blobs_[0]->cpu_data();
    const float* gv = blobs_[1]->cpu_data();
    for (int c = 0; c < C_; ++c) {
      const float inv = 1.f / std::sqrt(gv[c] + eps_);
      for (int n = 0; n < N; ++n) {
        const float* xp = x + ((long)n * C_ + c) * S;
        float* yp = y + ((long)n * C_ + c) * S;
        for (long s = 0; s < S; ++s) {
          const float v = (xp[s] - gm[c]) * inv;
          yp[s] = scale_bias_ ? v * sc[c] + bi[c] : v;
        }
      }
    }
    return;
  }
  float* mean = mean_.mutable_cpu_data();
  float* var = var_.mutable_cpu_data();
  float* inv = inv_std_.mutable_cpu_data();
#pragma omp parallel for schedule(static)
  for (int c = 0; c < C_; ++c) {
    double acc = 0;
    for (int n = 0; n < N; ++n) {
      const float* xp = x + ((long)n * C_ + c) * S;
      for (long s = 0; s < S; ++s) acc += xp[s];
    }
    mean[c] = (float)(acc / ((double)N * S));
    double v2 = 0;
    for (int n = 0; n < N; ++n) {
      const float* xp = x + ((long)n * C_ + c) * S;
      for (long s = 0; s < S; ++s) {
        const float d = xp[s] - mean[c];
        v2 += (double)d * d;
      }
    }
    var[c] = (float)(v2 / ((double)N * S));
    inv[c] = 1.f / std::sqrt(var[c] + eps_);
    for (int n = 0; n < N; ++n) {
      const float* xp = x + ((long)n * C_ + c) * S;
      float* yp = y + ((long)n * C_ + c) * S;
      for (long s = 0; s < S; ++s) {
        const float v = (xp[s] - mean[c]) * inv[c];
        yp[s] = scale_bias_ ? v * sc[c] + bi[c] : v;
      }
    }
  }
  // moving averages: copies on the first TWO forwards, then blends
  // (batch_norm_layer.cpp:200-213 'iter_ > 1' semantics); per-GPU, never
  // synchronized across ranks
  float* gm = blobs_[0]->mutable_cpu_data();
  float* gv = blobs_[1]->mutable_cpu_data();
  if (iter_ > 1) {
    for (int c = 0; c < C_; ++c) {
      gm[c] = (1.f - maf_) * mean[c] + maf_ * gm[c];
      gv[c] = (1.f - maf_) * var[c] + maf_ * gv[c];
    }
  } else {
    memcpy(gm, mean, sizeof(float) * C_);
    memcpy(gv, var, sizeof(float) * C_);
  }
  ++iter_;
}

void BatchNormLayer::Backward_cpu(const std::vector<Blob*>& top,
                                  const std::vector<bool>& prop_down,
                                  const std::vector<Blob*>& bottom) {
  const int N = bottom[0]->num();
  const long S = bottom[0]->count() / ((long)N * C_);
  const float* x = bwd_x(bottom, top, false);  // saved copy when in-place
  const float* dy = top[0]->cpu_diff();
  const float* mean = mean_.cpu_data();
  const float* inv = inv_std_.cpu_data();
  const float* sc = scale_bias_ ? blobs_[3]->cpu_data() : nullptr;
  float* dx = prop_down[0] ? bottom[0]->mutable_cpu_diff() : nullptr;
  // hoist the lazy head-state transitions out of the parallel region
  // (mutable_cpu_diff inside the omp loop raced on first allocation)
  float* dscale = scale_bias_ ? blobs_[3]->mutable_cpu_diff() : nullptr;
  float* dbias = scale_bias_ ? blobs_[4]->mutable_cpu_diff() : nullptr;
#pragma omp parallel for schedule(static)
  for (int c = 0; c < C_; ++c) {
    double s_dy = 0, s_dyxn = 0;
    for (int n = 0; n < N; ++n) {
      const float* dyp = dy + ((long)n * C_ + c) * S;
      const float* xp = x + ((long)n * C_ + c) * S;
      for (long s = 0; s < S; ++s) {
        const float xn = (xp[s] - mean[c]) * inv[c];
        s_dy += dyp[s];
        s_dyxn += (double)dyp[s] * xn;
      }
    }
    if (scale_bias_) {
      dscale[c] = (float)s_dyxn;
      dbias[c] = (float)s_dy;
    }
    if (!dx) continue;
    const float scc = scale_bias_ ? sc[c] : 1.f;
    const float mdy = (float)(scc * s_dy / ((double)N * S));
    const float mdyxn = (float)(scc * s_dyxn / ((double)N * S));
    for (int n = 0; n < N; ++n) {
      const float* dyp = dy + ((long)n * C_ + c) * S;
      const float* xp = x + ((long)n * C_ + c) * S;
      float* dxp = dx + ((long)n * C_ + c) * S;
      for (long s = 0; s < S; ++s) {
        const float xn = (xp[s] - mean[c]) * inv[c];
        dxp[s] = (dyp[s] * scc - mdy - mdyxn * xn) * inv[c];
      }
    }
  }
}

// ---------------------------------------------------------------- ReLU
void ReLULayer::Forward_cpu(const std::vector<Blob*>& bottom,
                            const std::vector<Blob*>& top) {
  auto rp = param_->sub("relu_param");
  const float slope = rp ? (float)rp->num("negative_slope", 0) : 0.f;
  const float* x = bottom[0]->cpu_data();
  float* y = top[0]->mutable_cpu_data();
  const long n = bottom[0]->count();
#pragma omp parallel for schedule(static)
  for (long i = 0; i < n; ++i) y[i] = x[i] > 0 ? x[i] : slope * x[i];
}

void ReLULayer::Backward_cpu(const std::vector<Blob*>& top,
                             const std::vector<bool>& prop_down,
                             const std::vector<Blob*>& bottom) {
  if (!prop_down[0]) return;
  auto rp = param_->sub("relu_param");
  const float slope = rp ? (float)rp->num("negative_slope", 0) : 0.f;
  const float* x = bottom[0]->cpu_data();
  const float* dy = top[0]->cpu_diff();
  float* dx = bottom[0]->mutable_cpu_diff();
  const long n = bottom[0]->count();
#pragma omp parallel for schedule(static)
  for (long i = 0; i < n; ++i)
    dx[i] = dy[i] * ((x[i] > 0) + slope * (x[i] <= 0));
}

// ---------------------------------------------------------------- Eltwise
void EltwiseLayer::LayerSetUp(const std::vector<Blob*>& bottom,
                              const std::vector<Blob*>&) {
  auto ep = param_->sub("eltwise_param");
  op_ = ep ? ep->str("operation", "SUM") : "SUM";
  CHECK_(op_ == "SUM") << "only Eltwise SUM is on the hot path (got " << op_
                       << ")";
  coeffs_.assign(bottom.size(), 1.f);
  if (ep) {
    auto cs = ep->nums("coeff");
    for (size_t i = 0; i < cs.size() && i < coeffs_.size(); ++i)
      coeffs_[i] = (float)cs[i];
  }
}

void EltwiseLayer::Forward_cpu(const std::vector<Blob*>& bottom,
                               const std::vector<Blob*>& top) {
  const long n = top[0]->count();
  float* y = top[0]->mutable_cpu_data();
  memset(y, 0, sizeof(float) * n);
  for (size_t i = 0; i < bottom.size(); ++i)
    cpu::axpy(n, coeffs_[i], bottom[i]->cpu_data(), y);
}

void EltwiseLayer::Backward_cpu(const std::vector<Blob*>& top,
                                const std::vector<bool>& prop_down,
                                const std::vector<Blob*>& bottom) {
  const long n = top[0]->count();
  const float* dy = top[0]->cpu_diff();
  for (size_t i = 0; i < bottom.size(); ++i) {
    if (!prop_down[i]) continue;
    float* dx = bottom[i]->mutable_cpu_diff();
    if (coeffs_[i] == 1.f)
      memcpy(dx, dy, sizeof(float) * n);
    else
      for (long j = 0; j < n; ++j) dx[j] = coeffs_[i] * dy[j];
  }
}

// ---------------------------------------------------------------- LRN
void LRNLayer::LayerSetUp(const std::vector<Blob*>&,
                          const std::vector<Blob*>&) {
  auto lp = param_->sub("lrn_param");
  size_ = lp ? (int)lp->inum("local_size", 5) : 5;
  CHECK_EQ_(size_ % 2, 1);
  alpha_ = lp ? (float)lp->num("alpha", 1.0) : 1.f;
  beta_ = lp ? (float)lp->num("beta", 0.75) : 0.75f;
  k_ = lp ? (float)lp->num("k", 1.0) : 1.f;
  const std::string region =
      lp ? lp->str("norm_region", "ACROSS_CHANNELS") : "ACROSS_CHANNELS";
  CHECK_(region == "ACROSS_CHANNELS");
}

void LRNLayer::Forward_cpu(const std::vector<Blob*>& bottom,
                           const std::vector<Blob*>& top) {
  const int N = bottom[0]->num(), C = bottom[0]->channels(),
            H = bottom[0]->height(), W = bottom[0]->width();
  const long S = (long)H * W;
  const int pre = (size_ - 1) / 2;
  const float aos = alpha_ / size_;
  const float* x = bottom[0]->cpu_data();
  float* sc = scale_.mutable_cpu_data();
  float* y = top[0]->mutable_cpu_data();
#pragma omp parallel for schedule(static)
  for (int n = 0; n < N; ++n)
    for (long s = 0; s < S; ++s)
      for (int c = 0; c < C; ++c) {
        float acc = 0;
        for (int cc = c - pre; cc <= c - pre + size_ - 1; ++cc)
          if (cc >= 0 && cc < C) {
            const float v = x[((long)n * C + cc) * S + s];
            acc += v * v;
          }
        sc[((long)n * C + c) * S + s] = k_ + aos * acc;
      }
  const long total = bottom[0]->count();
#pragma omp parallel for schedule(static)
  for (long i = 0; i < total; ++i) y[i] = x[i] * std::pow(sc[i], -beta_);
}

void LRNLayer::Backward_cpu(const std::vector<Blob*>& top,
                            const std::vector<bool>& prop_down,
                            const std::vector<Blob*>& bottom) {
  if (!prop_down[0]) return;
  const int N = bottom[0]->num(), C = bottom[0]->channels(),
            H = bottom[0]->height(), W = bottom[0]->width();
  const long S = (long)H * W;
  const int pre = (size_ - 1) / 2;
  const float cr = 2.f * alpha_ * beta_ / size_;
  const float* x = bottom[0]->cpu_data();
  const float* y = top[0]->cpu_data();
  const float* dy = top[0]->cpu_diff();
  const float* sc = scale_.cpu_data();
  float* dx = bottom[0]->mutable_cpu_diff();
#pragma omp parallel for schedule(static)
  for (int n = 0; n < N; ++n)
    for (long s = 0; s < S; ++s)
      for (int c = 0; c < C; ++c) {
        float acc = 0;
        for (int cc = c - (size_ - 1 - pre); cc <= c + pre; ++cc)
          if (cc >= 0 && cc < C) {
            const long i = ((long)n * C + cc) * S + s;
            acc += dy[i] * y[i] / sc[i];
          }
        const long i = ((long)n * C + c) * S + s;
        dx[i] = dy[i] * std::pow(sc[i], -beta_) - cr * x[i] * acc;
      }
}

// ---------------------------------------------------------------- Dropout
void DropoutLayer::LayerSetUp(const std::vector<Blob*>&,
                              const std::vector<Blob*>&) {
  auto dp = param_->sub("dropout_param");
  ratio_ = dp ? (float)dp->num("dropout_ratio", 0.5) : 0.5f;
  scale_ = 1.f / (1.f - ratio_);
}

void DropoutLayer::Forward_cpu(const std::vector<Blob*>& bottom,
                               const std::vector<Blob*>& top) {
  const long n = bottom[0]->count();
  const float* x = bottom[0]->cpu_data();
  float* y = top[0]->mutable_cpu_data();
  if (phase_ != Phase::TRAIN) {
    if (x != y) memcpy(y, x, sizeof(float) * n);
    return;
  }
  uint8_t* mask = (uint8_t*)mask_.mutable_cpu_data();
  Engine& E = Engine::get();
  const uint64_t key =
      h_splitmix64(E.seed ^ 0xD0D0ull ^ ((uint64_t)E.rank << 40) ^
                   E.data_iter);
  for (long i = 0; i < n; ++i) {
    mask[i] = h_u01(h_splitmix64(key ^ (uint64_t)i)) >= ratio_;
    y[i] = x[i] * mask[i] * scale_;
  }
  ++iter_;
}

void DropoutLayer::Backward_cpu(const std::vector<Blob*>& top,
                                const std::vector<bool>& prop_down,
                                const std::vector<Blob*>& bottom) {
  if (!prop_down[0]) return;
  const long n = bottom[0]->count();
  const float* dy = top[0]->cpu_diff();
  float* dx = bottom[0]->mutable_cpu_diff();
  if (phase_ != Phase::TRAIN) {
    if (dy != dx) memcpy(dx, dy, sizeof(float) * n);
    return;
  }
  const uint8_t* mask = (const uint8_t*)mask_.cpu_data();
  for (long i = 0; i < n; ++i) dx[i] = dy[i] * mask[i] * scale_;
}

// ---------------------------------------------------------------- Concat
void ConcatLayer::Reshape(const std::vector<Blob*>& bottom,
                          const std::vector<Blob*>& top) {
  // channel concat only (axis 1, the GoogLeNet/inception form): any other
  // concat_param.axis must fail loudly, not silently concat channels
  if (auto cp = param_->sub("concat_param")) {
    CHECK_EQ_(cp->inum("axis", 1), 1)
        << "only channel concat (axis 1) is implemented";
    CHECK_(!cp->has("concat_dim") || cp->inum("concat_dim") == 1)
        << "only channel concat (concat_dim 1) is implemented";
  }
  int C = 0;
  for (auto* b : bottom) {
    // every non-concat axis must match (reference concat_layer.cpp:46-52)
    CHECK_EQ_(b->num(), bottom[0]->num())
        << "Concat bottoms disagree on num";
    CHECK_EQ_(b->height(), bottom[0]->height())
        << "Concat bottoms disagree on height";
    CHECK_EQ_(b->width(), bottom[0]->width())
        << "Concat bottoms disagree on width";
    C += b->channels();
  }
  top[0]->Reshape({bottom[0]->num(), C, bottom[0]->height(),
                   bottom[0]->width()});
}

void ConcatLayer::Forward_cpu(const std::vector<Blob*>& bottom,
                              const std::vector<Blob*>& top) {
  const int N = top[0]->num(), Cd = top[0]->channels();
  const long S = top[0]->count(2);
  float* y = top[0]->mutable_cpu_data();
  int off = 0;
  for (auto* b : bottom) {
    const int Cs = b->channels();
    const float* x = b->cpu_data();
    for (int n = 0; n < N; ++n)
      memcpy(y + ((long)n * Cd + off) * S, x + (long)n * Cs * S,
             sizeof(float) * Cs * S);
    off += Cs;
  }
}

void ConcatLayer::Backward_cpu(const std::vector<Blob*>& top,
                               const std::vector<bool>& prop_down,
                               const std::vector<Blob*>& bottom) {
  const int N = top[0]->num(), Cd = top[0]->channels();
  const long S = top[0]->count(2);
  const float* dy = top[0]->cpu_diff();
  int off = 0;
  for (size_t i = 0; i < bottom.size(); ++i) {
    const int Cs = bottom[i]->channels();
    if (prop_down[i]) {
      float* dx = bottom[i]->mutable_cpu_diff();
      for (int n = 0; n < N; ++n)
        memcpy(dx + (long)n * Cs * S, dy + ((long)n * Cd + off) * S,
               sizeof(float) * Cs * S);
    }
    off += Cs;
  }
}

// ---------------------------------------------------------------- Split
void SplitLayer::Reshape(const std::vector<Blob*>& bottom,
                         const std::vector<Blob*>& top) {
  for (auto* t : top) {
    t->ReshapeLike(*bottom[0]);
    t->ShareData(*bottom[0]);
  }
}

void SplitLayer::Forward_cpu(const std::vector<Blob*>&,
                             const std::vector<Blob*>&) {}

void SplitLayer::Backward_cpu(const std::vector<Blob*>& top,
                              const std::vector<bool>& prop_down,
                              const std::vector<Blob*>& bottom) {
  if (!prop_down[0]) return;
  const long n = bottom[0]->count();
  float* dx = bottom[0]->mutable_cpu_diff();
  memcpy(dx, top[0]->cpu_diff(), sizeof(float) * n);
  for (size_t i = 1; i < top.size(); ++i)
    cpu::axpy(n, 1.f, top[i]->cpu_diff(), dx);
}

void SplitLayer::Backward_gpu(const std::vector<Blob*>& top,
                              const std::vector<bool>& prop_down,
                              const std::vector<Blob*>& bottom) {
  if (!prop_down[0]) return;
  Engine& E = Engine::get();
  const long n = bottom[0]->count();
  float* dx = bottom[0]->mutable_gpu_diff();
  if (top.size() == 1) {
    bottom[0]->ShareDiff(*top[0]);
    return;
  }
  if (top.size() == 2) {  // single fused pass for the common fan-out of 2
    gpu::add3(E.stream, n, top[0]->gpu_diff(), top[1]->gpu_diff(), dx);
    return;
  }
  gpu::copy(E.stream, n, top[0]->gpu_diff(), dx);
  for (size_t i = 1; i < top.size(); ++i)
    gpu::acc(E.stream, n, top[i]->gpu_diff(), dx);
}

// ---------------------------------------------------------------- Softmax
void SoftmaxLayer::Reshape(const std::vector<Blob*>& bottom,
                           const std::vector<Blob*>& top) {
  if (auto sp = param_->sub("softmax_param"))
    CHECK_EQ_(sp->inum("axis", 1), 1)
        << "only softmax over axis 1 (channels) is implemented";
  top[0]->ReshapeLike(*bottom[0]);
  outer_ = bottom[0]->num();
  C_ = bottom[0]->channels();
  inner_ = (int)bottom[0]->count(2);
}

static void softmax_cpu_impl(const float* x, int outer, int C, int inner,
                             float* y) {
#pragma omp parallel for collapse(2) schedule(static)
  for (int o = 0; o < outer; ++o)
    for (int s = 0; s < inner; ++s) {
      const float* xp = x + (long)o * C * inner + s;
      float* yp = y + (long)o * C * inner + s;
      float mx = xp[0];
      for (int c = 1; c < C; ++c) mx = std::max(mx, xp[(long)c * inner]);
      float sum = 0;
      for (int c = 0; c < C; ++c) {
        const float e = std::exp(xp[(long)c * inner] - mx);
        yp[(long)c * inner] = e;
        sum += e;
      }
      for (int c = 0; c < C; ++c) yp[(long)c * inner] /= sum;
    }
}

void SoftmaxLayer::Forward_cpu(const std::vector<Blob*>& bottom,
                               const std::vector<Blob*>& top) {
  softmax_cpu_impl(bottom[0]->cpu_data(), outer_, C_, inner_,
                   top[0]->mutable_cpu_data());
}

void SoftmaxLayer::Backward_cpu(const std::vector<Blob*>& top,
                                const std::vector<bool>& prop_down,
                                const std::vector<Blob*>& bottom) {
  if (!prop_down[0]) return;
  const float* y = top[0]->cpu_data();
  const float* dy = top[0]->cpu_diff();
  float* dx = bottom[0]->mutable_cpu_diff();
  for (int o = 0; o < outer_; ++o)
    for (int s = 0; s < inner_; ++s) {
      float dot = 0;
      for (int c = 0; c < C_; ++c) {
        const long i = ((long)o * C_ + c) * inner_ + s;
        dot += dy[i] * y[i];
      }
      for (int c = 0; c < C_; ++c) {
        const long i = ((long)o * C_ + c) * inner_ + s;
        dx[i] = (dy[i] - dot) * y[i];
      }
    }
}

// ---------------------------------------------------------- SoftmaxWithLoss
void SoftmaxWithLossLayer::LayerSetUp(const std::vector<Blob*>&,
                                      const std::vector<Blob*>&) {}

void SoftmaxWithLossLayer::Reshape(const std::vector<Blob*>& bottom,
                                   const std::vector<Blob*>& top) {
  outer_ = bottom[0]->num();
  C_ = bottom[0]->channels();
  inner_ = (int)bottom[0]->count(2);
  // one label per prediction row (reference softmax_loss_layer.cpp:57)
  CHECK_EQ_(bottom[1]->count(), (long)outer_ * inner_)
      << "SoftmaxWithLoss label count must equal outer*inner";
  prob_.ReshapeLike(*bottom[0]);
  top[0]->Reshape({1});
}

void SoftmaxWithLossLayer::Forward_cpu(const std::vector<Blob*>& bottom,
                                       const std::vector<Blob*>& top) {
  softmax_cpu_impl(bottom[0]->cpu_data(), outer_, C_, inner_,
                   prob_.mutable_cpu_data());
  const float* prob = prob_.cpu_data();
  const float* label = bottom[1]->cpu_data();
  double loss = 0;
  for (int o = 0; o < outer_; ++o)
    for (int s = 0; s < inner_; ++s) {
      const int lv = (int)label[(long)o * inner_ + s];
      const float p = prob[((long)o * C_ + lv) * inner_ + s];
      loss -= std::log(std::max(p, 1.175494e-38f));
    }
  top[0]->mutable_cpu_data()[0] =
      (float)(loss / ((double)outer_ * inner_));
}

void SoftmaxWithLossLayer::Backward_cpu(const std::vector<Blob*>& top,
                                        const std::vector<bool>& prop_down,
                                        const std::vector<Blob*>& bottom) {
  if (!prop_down[0]) return;
  const float lw = loss(0);
  const float w = lw / ((float)outer_ * inner_);
  const float* prob = prob_.cpu_data();
  const float* label = bottom[1]->cpu_data();
  float* dx = bottom[0]->mutable_cpu_diff();
  const long n = (long)outer_ * C_ * inner_;
  memcpy(dx, prob, sizeof(float) * n);
  for (int o = 0; o < outer_; ++o)
    for (int s = 0; s < inner_; ++s) {
      const int lv = (int)label[(long)o * inner_ + s];
      dx[((long)o * C_ + lv) * inner_ + s] -= 1.f;
    }
  for (long i = 0; i < n; ++i) dx[i] *= w;
  (void)top;
}

// ---------------------------------------------------------------- Accuracy
void AccuracyLayer::Reshape(const std::vector<Blob*>& bottom,
                            const std::vector<Blob*>& top) {
  auto ap = param_->sub("accuracy_param");
  top_k_ = ap ? (int)ap->inum("top_k", 1) : 1;
  CHECK_EQ_(bottom[1]->count(),
            (long)bottom[0]->num() * bottom[0]->count(2))
      << "Accuracy label count must equal outer*inner";
  top[0]->Reshape({1});
}

void AccuracyLayer::Forward_cpu(const std::vector<Blob*>& bottom,
                                const std::vector<Blob*>& top) {
  const int outer = bottom[0]->num();
  const int C = bottom[0]->channels();
  const int inner = (int)bottom[0]->count(2);
  const float* pred = bottom[0]->cpu_data();
  const float* label = bottom[1]->cpu_data();
  long correct = 0, total = 0;
  for (int o = 0; o < outer; ++o)
    for (int s = 0; s < inner; ++s) {
      const int lv = (int)label[(long)o * inner + s];
      const float pv = pred[((long)o * C + lv) * inner + s];
      ++total;
      // NaN predictions count as INCORRECT: the strictly-greater rank test
      // is vacuously 0 for NaN and would report accuracy 1.0 on a fully
      // divergent (all-NaN) net — the reference shares the rank form but an
      // engine must not present NaN as a perfect score
      if (pv != pv) continue;
      int rank = 0;
      for (int c = 0; c < C; ++c)
        if (pred[((long)o * C + c) * inner + s] > pv) ++rank;
      if (rank < top_k_) ++correct;
    }
  top[0]->mutable_cpu_data()[0] = total ? (float)correct / total : 0.f;
}

// ----------------------------------------------------------------- Input
// reference InputLayer: tops shaped by input_param, filled externally
class InputLayer : public Layer {
 public:
  using Layer::Layer;
  int min_bottom_blobs() const override { return 0; }
  int max_bottom_blobs() const override { return 0; }
  int max_top_blobs() const override { return 4096; }
  void Reshape(const std::vector<Blob*>&,
               const std::vector<Blob*>& top) override {
    auto ip = param_->sub("input_param");
    CHECK_(ip) << "Input layer needs input_param";
    auto shapes = ip->subs("shape");
    CHECK_EQ_((int)shapes.size(), (int)top.size());
    for (size_t i = 0; i < top.size(); ++i) {
      std::vector<int> dims;
      for (long d : shapes[i]->inums("dim")) dims.push_back((int)d);
      if (top[i]->shape() != dims) top[i]->Reshape(dims);
    }
  }
  void Forward_cpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override {}
  void Forward_gpu(const std::vector<Blob*>&,
                   const std::vector<Blob*>&) override {}
  void Backward_cpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override {}
  void Backward_gpu(const std::vector<Blob*>&, const std::vector<bool>&,
                    const std::vector<Blob*>&) override {}
};

// registry entries
REGISTER_LAYER("Input", InputLayer)
REGISTER_LAYER("Data", DataLayer)
REGISTER_LAYER("DummyData", DataLayer)
REGISTER_LAYER("Convolution", ConvolutionLayer)
REGISTER_LAYER("InnerProduct", InnerProductLayer)
REGISTER_LAYER("Pooling", PoolingLayer)
// ---------------------------------------------------------------- Scale
// Reference layers/scale_layer.cpp (+ bias inside, :121-134): channel-wise
// y = x * scale[c] (+ bias[c]); the BVLC BatchNorm+Scale prototxt pattern.
// Single-bottom learnable form only (axis 1, num_axes 1 — what every
// public BVLC-format ResNet uses); the two-bottom form is unsupported.
void ScaleLayer::LayerSetUp(const std::vector<Blob*>& bottom,
                            const std::vector<Blob*>&) {
  CHECK_EQ_((long)bottom.size(), 1)
      << "Scale: only the single-bottom learnable form is supported";
  auto sp = param_->sub("scale_param");
  const long axis = sp ? sp->inum("axis", 1) : 1;
  const long num_axes = sp ? sp->inum("num_axes", 1) : 1;
  CHECK_EQ_(axis, 1);
  CHECK_EQ_(num_axes, 1);
  bias_ = sp && sp->boolean("bias_term", false);
  C_ = bottom[0]->channels();
  if (blobs_.empty()) {
    Engine& E = Engine::get();
    blobs_.emplace_back(new Blob({C_}));
    auto filler = sp ? sp->sub("filler") : nullptr;
    if (filler) {
      fill_blob(*blobs_[0], filler, E.cpu_rng);
    } else {  // reference default: scale = 1 (scale_layer.cpp:39-44)
      float* p = blobs_[0]->mutable_cpu_data();
      for (int c = 0; c < C_; ++c) p[c] = 1.f;
    }
    if (bias_) {
      blobs_.emplace_back(new Blob({C_}));
      fill_blob(*blobs_[1], sp->sub("bias_filler"), E.cpu_rng);
    }
  }
}

void ScaleLayer::Forward_cpu(const std::vector<Blob*>& bottom,
                             const std::vector<Blob*>& top) {
  const int N = bottom[0]->num();
  const long S = bottom[0]->count() / ((long)N * C_);
  if (bottom[0] == top[0]) {  // in-place: keep x for backward
    temp_.ReshapeLike(*bottom[0]);
    memcpy(temp_.mutable_cpu_data(), bottom[0]->cpu_data(),
           sizeof(float) * bottom[0]->count());
  }
  const float* x = bottom[0]->cpu_data();
  const float* sc = blobs_[0]->cpu_data();
  const float* bi = bias_ ? blobs_[1]->cpu_data() : nullptr;
  float* y = top[0]->mutable_cpu_data();
  for (int n = 0; n < N; ++n)
    for (int c = 0; c < C_; ++c) {
      const float s = sc[c], b = bi ? bi[c] : 0.f;
      const float* xp = x + ((long)n * C_ + c) * S;
      float* yp = y + ((long)n * C_ + c) * S;
      for (long i = 0; i < S; ++i) yp[i] = xp[i] * s + b;
    }
}

void ScaleLayer::Backward_cpu(const std::vector<Blob*>& top,
                              const std::vector<bool>& prop_down,
                              const std::vector<Blob*>& bottom) {
  const int N = bottom[0]->num();
  const long S = bottom[0]->count() / ((long)N * C_);
  const float* dy = top[0]->cpu_diff();
  const float* x = bottom[0] == top[0] ? temp_.cpu_data()
                                       : bottom[0]->cpu_data();
  const float* sc = blobs_[0]->cpu_data();
  float* dsc = blobs_[0]->mutable_cpu_diff();
  float* dbi = bias_ ? blobs_[1]->mutable_cpu_diff() : nullptr;
  // in-place safe: param sums read (x, dy) before dx overwrites
  std::vector<double> s_dx(C_, 0.0), s_dy(C_, 0.0);
  for (int n = 0; n < N; ++n)
    for (int c = 0; c < C_; ++c) {
      const float* xp = x + ((long)n * C_ + c) * S;
      const float* dp = dy + ((long)n * C_ + c) * S;
      double a = 0, b = 0;
      for (long i = 0; i < S; ++i) {
        a += (double)dp[i] * xp[i];
        b += dp[i];
      }
      s_dx[c] += a;
      s_dy[c] += b;
    }
  for (int c = 0; c < C_; ++c) {
    dsc[c] = (float)s_dx[c];
    if (dbi) dbi[c] = (float)s_dy[c];
  }
  if (prop_down[0]) {
    float* dx = bottom[0]->mutable_cpu_diff();
    for (int n = 0; n < N; ++n)
      for (int c = 0; c < C_; ++c) {
        const float s = sc[c];
        const float* dp = dy + ((long)n * C_ + c) * S;
        float* op = dx + ((long)n * C_ + c) * S;
        for (long i = 0; i < S; ++i) op[i] = dp[i] * s;
      }
  }
}

// ----------------------------------------------------------------- Bias
// Reference layers/bias_layer.cpp: y = x + bias[c] (channel axis,
// learnable single-bottom form)
void BiasLayer::LayerSetUp(const std::vector<Blob*>& bottom,
                           const std::vector<Blob*>&) {
  CHECK_EQ_((long)bottom.size(), 1)
      << "Bias: only the single-bottom learnable form is supported";
  auto bp = param_->sub("bias_param");
  CHECK_EQ_(bp ? bp->inum("axis", 1) : 1, 1);
  CHECK_EQ_(bp ? bp->inum("num_axes", 1) : 1, 1);
  C_ = bottom[0]->channels();
  if (blobs_.empty()) {
    blobs_.emplace_back(new Blob({C_}));
    fill_blob(*blobs_[0], bp ? bp->sub("filler") : nullptr,
              Engine::get().cpu_rng);
  }
}

void BiasLayer::Forward_cpu(const std::vector<Blob*>& bottom,
                            const std::vector<Blob*>& top) {
  const int N = bottom[0]->num();
  const long S = bottom[0]->count() / ((long)N * C_);
  const float* x = bottom[0]->cpu_data();
  const float* bi = blobs_[0]->cpu_data();
  float* y = top[0]->mutable_cpu_data();
  for (int n = 0; n < N; ++n)
    for (int c = 0; c < C_; ++c) {
      const float b = bi[c];
      const float* xp = x + ((long)n * C_ + c) * S;
      float* yp = y + ((long)n * C_ + c) * S;
      for (long i = 0; i < S; ++i) yp[i] = xp[i] + b;
    }
}

void BiasLayer::Backward_cpu(const std::vector<Blob*>& top,
                             const std::vector<bool>& prop_down,
                             const std::vector<Blob*>& bottom) {
  const int N = bottom[0]->num();
  const long S = bottom[0]->count() / ((long)N * C_);
  const float* dy = top[0]->cpu_diff();
  float* dbi = blobs_[0]->mutable_cpu_diff();
  for (int c = 0; c < C_; ++c) {
    double acc = 0;
    for (int n = 0; n < N; ++n) {
      const float* dp = dy + ((long)n * C_ + c) * S;
      for (long i = 0; i < S; ++i) acc += dp[i];
    }
    dbi[c] = (float)acc;
  }
  if (prop_down[0]) {
    float* dx = bottom[0]->mutable_cpu_diff();
    if (dx != dy)
      memcpy(dx, dy, sizeof(float) * bottom[0]->count());
  }
}

REGISTER_LAYER("BatchNorm", BatchNormLayer)
REGISTER_LAYER("Scale", ScaleLayer)
REGISTER_LAYER("Bias", BiasLayer)
REGISTER_LAYER("ReLU", ReLULayer)
REGISTER_LAYER("Eltwise", EltwiseLayer)
REGISTER_LAYER("LRN", LRNLayer)
REGISTER_LAYER("Dropout", DropoutLayer)
REGISTER_LAYER("Concat", ConcatLayer)
REGISTER_LAYER("Split", SplitLayer)
REGISTER_LAYER("Softmax", SoftmaxLayer)
REGISTER_LAYER("SoftmaxWithLoss", SoftmaxWithLossLayer)
REGISTER_LAYER("Accuracy", AccuracyLayer)

}  // namespace camd
