// proto_wire.hpp — minimal protobuf (proto2) wire-format codec for the
// snapshot interop subset (no protoc in this environment):
//   .caffemodel  = NetParameter  { name=1, layer=100 (LayerParameter
//                  { name=1, type=2, blobs=7 (BlobProto) }) }
//   .solverstate = SolverState   { iter=1, learned_net=2, history=3
//                  (BlobProto), current_step=4 }
//   BlobProto    = { data=5 (packed float), shape=7 (BlobShape
//                  { dim=1 (packed int64) }) }
// Field numbers cited from the reference schema
// (src/caffe/proto/caffe.proto:15-40, :88-117, :303-308, :368-399).
#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace camd {
namespace wire {

// ------------------------------------------------------------- writer
class Writer {
 public:
  std::string out;

  void varint(uint64_t v) {
    while (v >= 0x80) {
      out.push_back((char)(v | 0x80));
      v >>= 7;
    }
    out.push_back((char)v);
  }
  void tag(int field, int wt) { varint(((uint64_t)field << 3) | wt); }
  void str(int field, const std::string& s) {
    tag(field, 2);
    varint(s.size());
    out += s;
  }
  void vint(int field, int64_t v) {
    tag(field, 0);
    varint((uint64_t)v);
  }
  void packed_floats(int field, const float* p, long n) {
    tag(field, 2);
    varint((uint64_t)n * 4);
    out.append((const char*)p, n * 4);
  }
  void packed_i64(int field, const std::vector<int64_t>& v) {
    Writer tmp;
    for (int64_t d : v) tmp.varint((uint64_t)d);
    tag(field, 2);
    varint(tmp.out.size());
    out += tmp.out;
  }
  void submsg(int field, const std::string& bytes) {
    tag(field, 2);
    varint(bytes.size());
    out += bytes;
  }
};

// ------------------------------------------------------------- reader
struct Field {
  int num;
  int wt;
  uint64_t vint;        // wt 0
  const char* data;     // wt 2
  size_t len;           // wt 2
};

class Reader {
 public:
  Reader(const char* p, size_t n) : p_(p), end_(p + n) {}
  bool next(Field* f);

 private:
  uint64_t varint();
  const char* p_;
  const char* end_;
  bool ok_ = true;

 public:
  bool ok() const { return ok_; }
};

// blob payload parsed from a BlobProto submessage
struct BlobData {
  std::vector<int> shape;
  std::vector<float> data;
};
BlobData parse_blob(const char* p, size_t n);

}  // namespace wire
}  // namespace camd
