#include "proto.hpp"

#include <fstream>

namespace camd {

namespace {

struct Lexer {
  const std::string& s;
  size_t i = 0;
  explicit Lexer(const std::string& t) : s(t) {}

  void skip_ws() {
    while (i < s.size()) {
      char c = s[i];
      if (c == '#') {
        while (i < s.size() && s[i] != '\n') ++i;
      } else if (c == ' ' || c == '\t' || c == '\n' || c == '\r' ||
                 c == ',' || c == ';') {
        ++i;
      } else {
        break;
      }
    }
  }

  bool eof() {
    skip_ws();
    return i >= s.size();
  }

  char peek() {
    skip_ws();
    return i < s.size() ? s[i] : '\0';
  }

  std::string token() {
    skip_ws();
    CHECK_(i < s.size()) << "unexpected EOF in prototxt";
    char c = s[i];
    if (c == '{' || c == '}' || c == ':' || c == '<' || c == '>') {
      ++i;
      return std::string(1, c);
    }
    if (c == '"' || c == '\'') {
      const char q = c;
      ++i;
      std::string out;
      while (i < s.size() && s[i] != q) out.push_back(s[i++]);
      CHECK_(i < s.size()) << "unterminated string in prototxt";
      ++i;
      return std::string("\"") + out;  // marker: quoted
    }
    std::string out;
    while (i < s.size()) {
      c = s[i];
      if (c == ' ' || c == '\t' || c == '\n' || c == '\r' || c == ':' ||
          c == '{' || c == '}' || c == '#' || c == ',' || c == ';' ||
          c == '<' || c == '>')
        break;
      out.push_back(c);
      ++i;
    }
    CHECK_(!out.empty()) << "empty token at pos " << i;
    return out;
  }
};

PMsgPtr parse_body(Lexer& lx, bool top) {
  auto msg = std::make_shared<PMsg>();
  while (true) {
    if (lx.eof()) {
      CHECK_(top) << "unexpected EOF inside message";
      return msg;
    }
    if (lx.peek() == '}' || lx.peek() == '>') {
      lx.token();
      CHECK_(!top) << "unmatched } at top level";
      return msg;
    }
    std::string name = lx.token();
    PVal v;
    char p = lx.peek();
    if (p == ':') {
      lx.token();
      if (lx.peek() == '{' || lx.peek() == '<') {  // name: { ... } also legal
        lx.token();
        v.kind = PVal::MSG;
        v.msg = parse_body(lx, false);
      } else {
        std::string val = lx.token();
        if (!val.empty() && val[0] == '"') val = val.substr(1);
        v.kind = PVal::SCALAR;
        v.scalar = val;
      }
    } else if (p == '{' || p == '<') {
      lx.token();
      v.kind = PVal::MSG;
      v.msg = parse_body(lx, false);
    } else {
      CAMD_FATAL << "expected ':' or '{' after field '" << name << "'";
    }
    msg->fields.emplace_back(name, std::move(v));
  }
}

}  // namespace

PMsgPtr parse_prototxt(const std::string& text) {
  Lexer lx(text);
  return parse_body(lx, true);
}

PMsgPtr parse_prototxt_file(const std::string& path) {
  std::ifstream f(path);
  CHECK_(f.good()) << "cannot open prototxt: " << path;
  std::string text((std::istreambuf_iterator<char>(f)),
                   std::istreambuf_iterator<char>());
  return parse_prototxt(text);
}

}  // namespace camd
