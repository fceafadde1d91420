#include "proto_wire.hpp"

#include <cstring>

namespace camd {
namespace wire {

uint64_t Reader::varint() {
  uint64_t v = 0;
  int shift = 0;
  while (p_ < end_) {
    const uint8_t b = (uint8_t)*p_++;
    v |= (uint64_t)(b & 0x7f) << shift;
    if (!(b & 0x80)) return v;
    shift += 7;
    if (shift > 63) break;
  }
  ok_ = false;
  return 0;
}

bool Reader::next(Field* f) {
  if (p_ >= end_ || !ok_) return false;
  const uint64_t key = varint();
  if (!ok_) return false;
  f->num = (int)(key >> 3);
  f->wt = (int)(key & 7);
  switch (f->wt) {
    case 0:
      f->vint = varint();
      return ok_;
    case 1:  // fixed64
      if (p_ + 8 > end_) return ok_ = false;
      f->data = p_;
      f->len = 8;
      p_ += 8;
      return true;
    case 2: {
      const uint64_t len = varint();
      if (!ok_ || p_ + len > end_) return ok_ = false;
      f->data = p_;
      f->len = (size_t)len;
      p_ += len;
      return true;
    }
    case 5:  // fixed32
      if (p_ + 4 > end_) return ok_ = false;
      f->data = p_;
      f->len = 4;
      p_ += 4;
      return true;
    default:
      ok_ = false;
      return false;
  }
}

BlobData parse_blob(const char* p, size_t n) {
  BlobData out;
  Reader r(p, n);
  Field f;
  while (r.next(&f)) {
    if (f.num == 5 && f.wt == 2) {  // packed float data
      const size_t cnt = f.len / 4;
      out.data.resize(cnt);
      memcpy(out.data.data(), f.data, cnt * 4);
    } else if (f.num == 5 && f.wt == 5) {  // unpacked float element
      float v;
      memcpy(&v, f.data, 4);
      out.data.push_back(v);
    } else if (f.num == 7 && f.wt == 2) {  // BlobShape
      Reader rs(f.data, f.len);
      Field fs;
      while (rs.next(&fs)) {
        if (fs.num == 1 && fs.wt == 2) {  // packed dims
          Reader rd(fs.data, fs.len);
          // packed varints: parse manually
          const char* q = fs.data;
          const char* qe = fs.data + fs.len;
          while (q < qe) {
            uint64_t v = 0;
            int shift = 0;
            while (q < qe) {
              const uint8_t b = (uint8_t)*q++;
              v |= (uint64_t)(b & 0x7f) << shift;
              if (!(b & 0x80)) break;
              shift += 7;
            }
            out.shape.push_back((int)v);
          }
          (void)rd;
        } else if (fs.num == 1 && fs.wt == 0) {
          out.shape.push_back((int)fs.vint);
        }
      }
    } else if (f.num >= 1 && f.num <= 4 && f.wt == 0 && out.shape.empty()) {
      // legacy num/channels/height/width — collect in order if present
      // (reference BlobProto:31-34); handled loosely: append
      out.shape.push_back((int)f.vint);
    }
  }
  return out;
}

}  // namespace wire
}  // namespace camd
