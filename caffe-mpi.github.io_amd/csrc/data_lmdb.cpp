// data_lmdb.cpp — the LMDB training-data feed:
// from-scratch LMDB reader (lmdb_reader.cpp) -> caffe::Datum wire decode
// -> rank-cycle sharding -> prefetch thread (double buffer, the
// reference's BasePrefetchingDataLayer/BlockingQueue design collapsed to
// one worker + two slots, base_data_layer.cpp:99-134) -> uint8 crop/
// mirror/mean/scale transform (CPU here; the GPU path uploads the uint8
// batch and runs k_transform, the data_transformer.cu:14-100 analog).
//
// Sharding (reference data_reader.cpp:295-301 semantics, one parser per
// rank): rank r of `world` reads record (iter*batch + j)*world + r mod N
// — ranks own disjoint record streams, deterministic and resumable from
// the solver iteration (the LMDB cursor-position analog; when N % world
// != 0 the wraparound rotates ownership across epochs, which keeps every
// record in use — the reference's full-cycle guarantee).
#include <condition_variable>
#include <functional>
#include <fstream>
#include <mutex>
#include <thread>

#include "layers.hpp"
#include "lmdb_reader.hpp"
#include "proto_wire.hpp"

namespace camd {

struct Datum {
  int c = 0, h = 0, w = 0;
  const uint8_t* data = nullptr;
  size_t len = 0;
  long label = 0;
};

// caffe.proto Datum: channels=1 height=2 width=3 data=4 label=5
// float_data=6 encoded=7
static Datum decode_datum(const uint8_t* p, size_t n) {
  Datum d;
  wire::Reader r((const char*)p, n);
  wire::Field f;
  while (r.next(&f)) {
    switch (f.num) {
      case 1: d.c = (int)f.vint; break;
      case 2: d.h = (int)f.vint; break;
      case 3: d.w = (int)f.vint; break;
      case 4:
        d.data = (const uint8_t*)f.data;
        d.len = f.len;
        break;
      case 5: d.label = (long)f.vint; break;
      case 7:
        CHECK_(f.vint == 0) << "encoded (JPEG) Datum records are not "
                               "supported (no image codec in this build)";
        break;
      default: break;
    }
  }
  CHECK_GT_(d.c, 0);
  CHECK_(d.data != nullptr)
      << "Datum record has no uint8 `data` field (float_data Datums are "
         "not supported by this feed)";
  CHECK_EQ_((long)d.len, (long)d.c * d.h * d.w)
      << "Datum data size mismatch";
  return d;
}

struct LmdbFeed {
  LmdbReader reader;
  // generalized source hooks (ImageData reuses the whole feed machinery):
  //  provider: fills one record's uint8 image + label (default: LMDB
  //    record at `idx` decoded as a Datum)
  //  index_of: maps (iter, j) to a record index (default: the rank-cycle
  //    shard; ImageData reads the FULL list per rank like the reference,
  //    with an optional per-epoch keyed shuffle)
  std::function<void(long idx, uint8_t* dst, float* label)> provider;
  std::function<long(uint64_t iter, int j)> index_of;
  long nrec = 0;
  int dc = 0, dh = 0, dw = 0;   // datum dims
  int crop = 0;                  // 0 = full image
  bool mirror = false;
  float scale = 1.f;
  std::vector<float> mean;       // per-channel (size dc) or per-pixel
  bool mean_per_pixel = false;
  int batch = 0, outH = 0, outW = 0;
  Phase phase = Phase::TRAIN;

  // prefetch double buffer (uint8 batch + labels + per-image geometry)
  struct Slot {
    std::vector<uint8_t> data;   // [batch][dc][dh][dw]
    std::vector<float> labels;   // [batch]
    std::vector<int> geo;        // [batch][3] = ho, wo, mirror
    uint64_t iter = ~0ull;
    bool ready = false;
  };
  Slot slots[2];
  std::mutex mu;
  std::condition_variable cv;
  std::thread worker;
  bool stop = false;
  uint64_t want = ~0ull;     // iteration the worker should fill next
  uint64_t filling = ~0ull;  // iteration the worker is CURRENTLY filling
                             // (guards get() from racing an in-flight
                             // fill on the same slot)

  // GPU upload staging: PINNED double buffers (parity = iter & 1) so the
  // H2D copies are truly async (no pageable staging), with an
  // event guarding reuse two iterations later.  The solver thread copies
  // slot -> pinned synchronously, so the worker's vectors are free the
  // moment get() returns.
  uint8_t* pin_u8[2] = {nullptr, nullptr};
  int* pin_geo[2] = {nullptr, nullptr};
  float* pin_lab[2] = {nullptr, nullptr};
  hipEvent_t ev[2] = {nullptr, nullptr};

  ~LmdbFeed() {
    {
      std::lock_guard<std::mutex> g(mu);
      stop = true;
    }
    cv.notify_all();
    if (worker.joinable()) worker.join();
    for (int p = 0; p < 2; ++p) {
      if (pin_u8[p]) (void)hipHostFree(pin_u8[p]);
      if (pin_geo[p]) (void)hipHostFree(pin_geo[p]);
      if (pin_lab[p]) (void)hipHostFree(pin_lab[p]);
      if (ev[p]) (void)hipEventDestroy(ev[p]);
    }
  }

  // deterministic crop/mirror stream keyed like the synthetic source
  void fill_slot(Slot& s, uint64_t iter) {
    Engine& E = Engine::get();
    const long imsz = (long)dc * dh * dw;
    s.data.resize((size_t)batch * imsz);
    s.labels.resize(batch);
    s.geo.assign((size_t)batch * 3, 0);
    const uint64_t key =
        h_splitmix64(E.seed ^ 0x17DBull ^ ((uint64_t)E.rank << 40) ^ iter);
    for (int j = 0; j < batch; ++j) {
      const long g =
          index_of ? index_of(iter, j)
                   : (long)((((iter * (uint64_t)batch + j) *
                              (uint64_t)E.world) +
                             E.rank) %
                            (uint64_t)nrec);
      if (provider) {
        provider(g, s.data.data() + (size_t)j * imsz, &s.labels[j]);
      } else {
        auto rec = reader.at(g);
        Datum d = decode_datum(rec.first, rec.second);
        CHECK_EQ_(d.c, dc);
        CHECK_EQ_(d.h, dh);
        CHECK_EQ_(d.w, dw);
        memcpy(s.data.data() + (size_t)j * imsz, d.data, imsz);
        s.labels[j] = (float)d.label;
      }
      int ho = 0, wo = 0, mir = 0;
      if (phase == Phase::TRAIN) {  // random crop + mirror (train)
        const uint64_t hj = h_splitmix64(key ^ (uint64_t)(3 * j + 1));
        const uint64_t wj = h_splitmix64(key ^ (uint64_t)(3 * j + 2));
        if (outH < dh) ho = (int)(hj % (uint64_t)(dh - outH + 1));
        if (outW < dw) wo = (int)(wj % (uint64_t)(dw - outW + 1));
        if (mirror)
          mir = (int)(h_splitmix64(key ^ (uint64_t)(3 * j + 3)) & 1);
      } else {  // center crop, no mirror (test)
        ho = (dh - outH) / 2;
        wo = (dw - outW) / 2;
      }
      s.geo[3 * j] = ho;
      s.geo[3 * j + 1] = wo;
      s.geo[3 * j + 2] = mir;
    }
    s.iter = iter;
    s.ready = true;
  }

  void start_worker() {
    worker = std::thread([this] {
      std::unique_lock<std::mutex> lk(mu);
      while (true) {
        cv.wait(lk, [this] { return stop || want != ~0ull; });
        if (stop) return;
        const uint64_t it = want;
        want = ~0ull;
        filling = it;
        Slot& s = slots[it & 1];
        lk.unlock();
        fill_slot(s, it);
        lk.lock();
        filling = ~0ull;
        cv.notify_all();
      }
    });
  }

  // get the decoded batch for `iter`: prefetched if the worker had it,
  // else decode inline; then kick the worker for iter+1
  Slot& get(uint64_t iter) {
    std::unique_lock<std::mutex> lk(mu);
    Slot& s = slots[iter & 1];
    // wait out the worker completely (assigned OR mid-fill): without the
    // `filling` guard an inline fill here could race an in-flight worker
    // fill of the SAME slot
    cv.wait(lk, [&] {
      return (want == ~0ull && filling == ~0ull) || stop;
    });
    if (!(s.ready && s.iter == iter)) {
      lk.unlock();
      fill_slot(s, iter);
      lk.lock();
    }
    s.ready = false;
    Slot& nxt = slots[(iter + 1) & 1];
    nxt.ready = false;
    want = iter + 1;
    cv.notify_all();
    return s;
  }

  // CPU transform: (uint8 - mean) * scale with crop/mirror — the
  // reference DataTransformer::Transform CPU path semantics
  void transform_cpu(const Slot& s, float* out) const {
    const long imsz = (long)dc * dh * dw;
#pragma omp parallel for schedule(static)
    for (int j = 0; j < batch; ++j) {
      const uint8_t* im = s.data.data() + (size_t)j * imsz;
      const int ho = s.geo[3 * j], wo = s.geo[3 * j + 1],
                mir = s.geo[3 * j + 2];
      float* op = out + (size_t)j * dc * outH * outW;
      for (int c = 0; c < dc; ++c)
        for (int y = 0; y < outH; ++y)
          for (int x = 0; x < outW; ++x) {
            const int sx = wo + x;  // mean subtract in SOURCE coords
            const float v = (float)im[((long)c * dh + ho + y) * dw + sx];
            const float m =
                mean.empty()
                    ? 0.f
                    : (mean_per_pixel
                           ? mean[((long)c * dh + ho + y) * dw + sx]
                           : mean[c]);
            const int ox = mir ? outW - 1 - x : x;
            op[((long)c * outH + y) * outW + ox] = (v - m) * scale;
          }
    }
  }
};

// shared transform_param parsing (Data + ImageData): crop/mirror/scale
// + mean_value or mean_file (binaryproto, per-pixel)
static void setup_transform(LmdbFeed& F, const PMsgPtr& tp);

void DataLayer::setup_lmdb(const std::string& source) {
  feed_.reset(new LmdbFeed());
  LmdbFeed& F = *feed_;
  F.reader.open(source);
  F.nrec = F.reader.size();
  CHECK_GT_(F.nrec, 0) << "empty LMDB " << source;
  auto rec0 = F.reader.at(0);
  Datum d0 = decode_datum(rec0.first, rec0.second);
  F.dc = d0.c;
  F.dh = d0.h;
  F.dw = d0.w;
  F.batch = batch_;
  F.phase = phase_;
  setup_transform(F, param_->sub("transform_param"));
  C_ = F.dc;
  H_ = F.outH;
  W_ = F.outW;
  F.start_worker();
}

static void setup_transform(LmdbFeed& F, const PMsgPtr& tp) {
  F.crop = tp ? (int)tp->inum("crop_size", 0) : 0;
  F.mirror = tp && tp->boolean("mirror", false);
  F.scale = tp ? (float)tp->num("scale", 1.0) : 1.f;
  F.outH = F.crop > 0 ? F.crop : F.dh;
  F.outW = F.crop > 0 ? F.crop : F.dw;
  CHECK_LE_(F.outH, F.dh);
  CHECK_LE_(F.outW, F.dw);
  if (tp && tp->has("mean_file")) {
    // binaryproto BlobProto mean (full-size, subtracted in source coords)
    std::ifstream f(tp->str("mean_file"), std::ios::binary);
    CHECK_(f.good()) << "cannot read mean_file " << tp->str("mean_file");
    std::string buf((std::istreambuf_iterator<char>(f)),
                    std::istreambuf_iterator<char>());
    auto blob = wire::parse_blob(buf.data(), buf.size());
    CHECK_EQ_((long)blob.data.size(), (long)F.dc * F.dh * F.dw)
        << "mean_file shape mismatch";
    F.mean = blob.data;
    F.mean_per_pixel = true;
  } else if (tp) {
    auto mv = tp->nums("mean_value");
    if (!mv.empty()) {
      F.mean.assign(F.dc, (float)mv[0]);
      for (int c = 0; c < F.dc && c < (int)mv.size(); ++c)
        F.mean[c] = (float)mv[c];
      F.mean_per_pixel = false;
    }
  }
}

void DataLayer::forward_lmdb_cpu(const std::vector<Blob*>& top) {
  Engine& E = Engine::get();
  LmdbFeed& F = *feed_;
  auto& slot = F.get(E.data_iter);
  F.transform_cpu(slot, top[0]->mutable_cpu_data());
  memcpy(top[1]->mutable_cpu_data(), slot.labels.data(),
         sizeof(float) * batch_);
}

void DataLayer::forward_lmdb_gpu(const std::vector<Blob*>& top) {
  Engine& E = Engine::get();
  LmdbFeed& F = *feed_;
  auto& slot = F.get(E.data_iter);
  const int p = (int)(E.data_iter & 1);
  const size_t usz = slot.data.size();
  if (!F.pin_u8[p]) {
    HIP_CHECK(hipHostMalloc((void**)&F.pin_u8[p], usz));
    HIP_CHECK(
        hipHostMalloc((void**)&F.pin_geo[p], sizeof(int) * slot.geo.size()));
    HIP_CHECK(hipHostMalloc((void**)&F.pin_lab[p], sizeof(float) * batch_));
    HIP_CHECK(hipEventCreateWithFlags(&F.ev[p], hipEventDisableTiming));
  } else {
    // the copies enqueued two iterations ago must have drained before the
    // pinned buffers are rewritten (GPU can run ~2 iters behind the host)
    HIP_CHECK(hipEventSynchronize(F.ev[p]));
  }
  memcpy(F.pin_u8[p], slot.data.data(), usz);
  memcpy(F.pin_geo[p], slot.geo.data(), sizeof(int) * slot.geo.size());
  memcpy(F.pin_lab[p], slot.labels.data(), sizeof(float) * batch_);
  Workspace& ws = Workspace::get_global();
  if (u8_slot_ < 0) {
    static int next_slot = 300;
    u8_slot_ = next_slot++;
    geo_slot_ = next_slot++;
    mean_slot_ = next_slot++;
  }
  uint8_t* du8 = (uint8_t*)ws.get(u8_slot_, usz);
  int* dgeo = (int*)ws.get(geo_slot_, sizeof(int) * slot.geo.size());
  HIP_CHECK(hipMemcpyAsync(du8, F.pin_u8[p], usz, hipMemcpyHostToDevice,
                           E.stream));
  HIP_CHECK(hipMemcpyAsync(dgeo, F.pin_geo[p],
                           sizeof(int) * slot.geo.size(),
                           hipMemcpyHostToDevice, E.stream));
  float* dmean = nullptr;
  if (!F.mean.empty()) {
    dmean = (float*)ws.get(mean_slot_, sizeof(float) * F.mean.size());
    if (!mean_uploaded_) {
      HIP_CHECK(hipMemcpy(dmean, F.mean.data(),
                          sizeof(float) * F.mean.size(),
                          hipMemcpyHostToDevice));
      mean_uploaded_ = true;
    }
  }
  gpu::transform_u8(E.stream, du8, batch_, F.dc, F.dh, F.dw, F.outH,
                    F.outW, dgeo, dmean, F.mean_per_pixel ? 2
                                         : dmean          ? 1
                                                          : 0,
                    F.scale, top[0]->mutable_gpu_data());
  HIP_CHECK(hipMemcpyAsync(top[1]->mutable_gpu_data(), F.pin_lab[p],
                           sizeof(float) * batch_, hipMemcpyHostToDevice,
                           E.stream));
  HIP_CHECK(hipEventRecord(F.ev[p], E.stream));
}


// ================================================================
// ImageData layer (reference src/caffe/layers/image_data_layer.cpp):
// image_data_param { source: listfile ("path label" per line),
// batch_size, new_height/new_width (bilinear resize), shuffle,
// root_folder } + the shared transform_param.  The image codecs are
// PPM (P6) and PGM (P5) — binary netpbm, the formats decodable without
// an image library (none exists in this environment; JPEG/PNG records
// fail with a clear message).  Like the reference, every rank reads the
// FULL list; shuffle is a per-epoch Fisher-Yates keyed by
// (seed, rank, epoch) so runs are reproducible and ranks decorrelate.
namespace {

struct PnmImage {
  int c = 0, h = 0, w = 0;
  std::vector<uint8_t> px;  // CHW
};

PnmImage load_pnm(const std::string& path) {
  std::ifstream f(path, std::ios::binary);
  CHECK_(f.good()) << "cannot open image " << path;
  std::string magic;
  f >> magic;
  CHECK_(magic == "P6" || magic == "P5")
      << "unsupported image format '" << magic << "' in " << path
      << " (only binary PPM/PGM decodable in this environment — "
         "no JPEG/PNG codec in the image)";
  auto next_int = [&]() {
    int v;
    // skip whitespace/comments (netpbm allows # comments in the header)
    while (true) {
      int ch = f.peek();
      if (ch == '#') {
        std::string line;
        std::getline(f, line);
      } else if (isspace(ch)) {
        f.get();
      } else {
        break;
      }
    }
    f >> v;
    return v;
  };
  PnmImage im;
  im.w = next_int();
  im.h = next_int();
  const int maxv = next_int();
  CHECK_EQ_(maxv, 255) << "only 8-bit netpbm supported: " << path;
  f.get();  // single whitespace before raster
  im.c = magic == "P6" ? 3 : 1;
  std::vector<uint8_t> raw((size_t)im.w * im.h * im.c);
  f.read((char*)raw.data(), (long)raw.size());
  CHECK_(f.good()) << "truncated image " << path;
  // interleaved HWC -> planar CHW (the engine/caffe layout)
  im.px.resize(raw.size());
  for (int y = 0; y < im.h; ++y)
    for (int x = 0; x < im.w; ++x)
      for (int ch = 0; ch < im.c; ++ch)
        im.px[((size_t)ch * im.h + y) * im.w + x] =
            raw[((size_t)y * im.w + x) * im.c + ch];
  return im;
}

// bilinear uint8 resize, CHW (the reference resizes via cv::resize
// INTER_LINEAR when new_height/new_width are set)
void resize_bilinear(const PnmImage& src, int nh, int nw,
                     std::vector<uint8_t>& dst) {
  dst.resize((size_t)src.c * nh * nw);
  const float sy = (float)src.h / nh, sx = (float)src.w / nw;
  for (int c = 0; c < src.c; ++c) {
    const uint8_t* sp = src.px.data() + (size_t)c * src.h * src.w;
    uint8_t* dp = dst.data() + (size_t)c * nh * nw;
    for (int y = 0; y < nh; ++y) {
      const float fy = (y + 0.5f) * sy - 0.5f;
      int y0 = (int)floorf(fy);
      const float wy = fy - y0;
      y0 = std::max(0, std::min(src.h - 1, y0));
      const int y1 = std::min(src.h - 1, y0 + 1);
      for (int x = 0; x < nw; ++x) {
        const float fx = (x + 0.5f) * sx - 0.5f;
        int x0 = (int)floorf(fx);
        const float wx = fx - x0;
        x0 = std::max(0, std::min(src.w - 1, x0));
        const int x1 = std::min(src.w - 1, x0 + 1);
        const float v =
            (1 - wy) * ((1 - wx) * sp[y0 * src.w + x0] +
                        wx * sp[y0 * src.w + x1]) +
            wy * ((1 - wx) * sp[y1 * src.w + x0] +
                  wx * sp[y1 * src.w + x1]);
        dp[y * nw + x] = (uint8_t)(v + 0.5f);
      }
    }
  }
}

}  // namespace

class ImageDataLayer : public DataLayer {
 public:
  using DataLayer::DataLayer;
  void LayerSetUp(const std::vector<Blob*>&,
                  const std::vector<Blob*>&) override {
    auto ip = param_->sub("image_data_param");
    CHECK_(ip) << "ImageData layer needs image_data_param";
    batch_ = (int)ip->inum("batch_size", 1);
    const std::string root = ip->str("root_folder", "");
    const std::string list = ip->str("source");
    std::ifstream f(list);
    CHECK_(f.good()) << "cannot read image list " << list;
    std::string path;
    long label;
    while (f >> path >> label) lines_.emplace_back(root + path, label);
    CHECK_GT_((long)lines_.size(), 0) << "empty image list " << list;
    nh_ = (int)ip->inum("new_height", 0);
    nw_ = (int)ip->inum("new_width", 0);
    shuffle_ = ip->boolean("shuffle", false);
    PnmImage first = load_pnm(lines_[0].first);
    const int dh = nh_ > 0 ? nh_ : first.h;
    const int dw = nw_ > 0 ? nw_ : first.w;

    feed_.reset(new LmdbFeed());
    LmdbFeed& F = *feed_;
    F.nrec = (long)lines_.size();
    F.dc = first.c;
    F.dh = dh;
    F.dw = dw;
    F.batch = batch_;
    F.phase = phase_;
    setup_transform(F, param_->sub("transform_param"));
    C_ = F.dc;
    H_ = F.outH;
    W_ = F.outW;
    // full list per rank (reference semantics); optional keyed shuffle
    F.index_of = [this](uint64_t iter, int j) {
      const uint64_t pos = iter * (uint64_t)batch_ + j;
      const long n = (long)lines_.size();
      const long epoch = (long)(pos / (uint64_t)n);
      const long i = (long)(pos % (uint64_t)n);
      if (!shuffle_) return i;
      if (epoch != perm_epoch_) reshuffle(epoch);
      return perm_[i];
    };
    F.provider = [this, &F](long idx, uint8_t* dst, float* label) {
      PnmImage im = load_pnm(lines_[idx].first);
      CHECK_EQ_(im.c, F.dc) << lines_[idx].first;
      if (nh_ > 0 || nw_ > 0) {
        std::vector<uint8_t> rs;
        resize_bilinear(im, F.dh, F.dw, rs);
        memcpy(dst, rs.data(), rs.size());
      } else {
        CHECK_EQ_(im.h, F.dh) << lines_[idx].first;
        CHECK_EQ_(im.w, F.dw) << lines_[idx].first;
        memcpy(dst, im.px.data(), im.px.size());
      }
      *label = (float)lines_[idx].second;
    };
    F.start_worker();
  }

 private:
  void reshuffle(long epoch) {
    Engine& E = Engine::get();
    const long n = (long)lines_.size();
    perm_.resize(n);
    for (long i = 0; i < n; ++i) perm_[i] = i;
    uint64_t key = h_splitmix64(E.seed ^ 0x1A6Eull ^
                                ((uint64_t)E.rank << 40) ^
                                (uint64_t)epoch);
    for (long i = n - 1; i > 0; --i) {  // keyed Fisher-Yates
      key = h_splitmix64(key);
      std::swap(perm_[i], perm_[key % (uint64_t)(i + 1)]);
    }
    perm_epoch_ = epoch;
  }
  std::vector<std::pair<std::string, long>> lines_;
  std::vector<long> perm_;
  long perm_epoch_ = -1;
  int nh_ = 0, nw_ = 0;
  bool shuffle_ = false;
};

REGISTER_LAYER("ImageData", ImageDataLayer)

}  // namespace camd
