// proto.hpp — hand-written protobuf text-format parser for the prototxt
// subset the four reference model sets use (SURVEY.md §8b grammar census;
// the reference's wire contract is src/caffe/proto/caffe.proto).  No protoc
// at runtime: a generic message tree + typed accessors with caffe.proto's
// defaults covers the ~40 fields needed.
#pragma once

#include <map>
#include <memory>
#include <string>
#include <vector>

#include "core.hpp"

namespace camd {

class PMsg;
using PMsgPtr = std::shared_ptr<PMsg>;

struct PVal {
  enum Kind { SCALAR, MSG } kind = SCALAR;
  std::string scalar;  // raw token (number, enum ident, quoted-stripped str)
  PMsgPtr msg;
};

class PMsg {
 public:
  // ordered (field, value) pairs; repeated fields appear multiple times
  std::vector<std::pair<std::string, PVal>> fields;

  bool has(const std::string& k) const {
    for (auto& f : fields)
      if (f.first == k) return true;
    return false;
  }
  const PVal* find(const std::string& k) const {
    for (auto& f : fields)
      if (f.first == k) return &f.second;
    return nullptr;
  }
  std::vector<const PVal*> all(const std::string& k) const {
    std::vector<const PVal*> v;
    for (auto& f : fields)
      if (f.first == k) v.push_back(&f.second);
    return v;
  }
  std::string str(const std::string& k, const std::string& dflt = "") const {
    auto* v = find(k);
    return v ? v->scalar : dflt;
  }
  double num(const std::string& k, double dflt = 0) const {
    auto* v = find(k);
    return v ? atof(v->scalar.c_str()) : dflt;
  }
  long inum(const std::string& k, long dflt = 0) const {
    auto* v = find(k);
    return v ? atol(v->scalar.c_str()) : dflt;
  }
  bool boolean(const std::string& k, bool dflt = false) const {
    auto* v = find(k);
    if (!v) return dflt;
    return v->scalar == "true" || v->scalar == "1";
  }
  PMsgPtr sub(const std::string& k) const {
    auto* v = find(k);
    return (v && v->kind == PVal::MSG) ? v->msg : nullptr;
  }
  std::vector<PMsgPtr> subs(const std::string& k) const {
    std::vector<PMsgPtr> out;
    for (auto* v : all(k))
      if (v->kind == PVal::MSG) out.push_back(v->msg);
    return out;
  }
  std::vector<long> inums(const std::string& k) const {
    std::vector<long> out;
    for (auto* v : all(k)) out.push_back(atol(v->scalar.c_str()));
    return out;
  }
  std::vector<double> nums(const std::string& k) const {
    std::vector<double> out;
    for (auto* v : all(k)) out.push_back(atof(v->scalar.c_str()));
    return out;
  }
  std::vector<std::string> strs(const std::string& k) const {
    std::vector<std::string> out;
    for (auto* v : all(k)) out.push_back(v->scalar);
    return out;
  }
};

// Parse protobuf text format. Throws on malformed input.
PMsgPtr parse_prototxt(const std::string& text);
PMsgPtr parse_prototxt_file(const std::string& path);

}  // namespace camd
