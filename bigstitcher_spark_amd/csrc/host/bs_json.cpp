#include "bs_json.h"

#include <cctype>
#include <cmath>
#include <cstring>
#include <cstdio>
#include <cstdlib>

namespace bsj {

namespace {
struct Parser {
  const char *p, *end;
  bool fail = false;
  void ws() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r'))
      ++p;
  }
  bool lit(const char *s) {
    size_t n = strlen(s);
    if ((size_t)(end - p) >= n && strncmp(p, s, n) == 0) {
      p += n;
      return true;
    }
    return false;
  }
  ValuePtr value() {
    ws();
    if (p >= end) {
      fail = true;
      return nullptr;
    }
    char c = *p;
    if (c == '{') return object();
    if (c == '[') return array();
    if (c == '"') {
      auto v = Value::mkstr("");
      if (!string(v->str)) return nullptr;
      return v;
    }
    if (lit("true")) return Value::mkbool(true);
    if (lit("false")) return Value::mkbool(false);
    if (lit("null")) return Value::mknull();
    return number();
  }
  bool string(std::string &out) {
    if (*p != '"') {
      fail = true;
      return false;
    }
    ++p;
    out.clear();
    while (p < end && *p != '"') {
      if (*p == '\\' && p + 1 < end) {
        ++p;
        switch (*p) {
          case 'n': out += '\n'; break;
          case 't': out += '\t'; break;
          case 'r': out += '\r'; break;
          case 'b': out += '\b'; break;
          case 'f': out += '\f'; break;
          case 'u': {
            if (end - p < 5) {
              fail = true;
              return false;
            }
            char buf[5] = {p[1], p[2], p[3], p[4], 0};
            unsigned code = (unsigned)strtoul(buf, nullptr, 16);
            /* BMP-only UTF-8 encode (metadata never needs more) */
            if (code < 0x80) {
              out += (char)code;
            } else if (code < 0x800) {
              out += (char)(0xC0 | (code >> 6));
              out += (char)(0x80 | (code & 0x3F));
            } else {
              out += (char)(0xE0 | (code >> 12));
              out += (char)(0x80 | ((code >> 6) & 0x3F));
              out += (char)(0x80 | (code & 0x3F));
            }
            p += 4;
            break;
          }
          default: out += *p;
        }
        ++p;
      } else {
        out += *p++;
      }
    }
    if (p >= end) {
      fail = true;
      return false;
    }
    ++p;
    return true;
  }
  ValuePtr number() {
    const char *s = p;
    if (p < end && (*p == '-' || *p == '+')) ++p;
    bool isint = true;
    while (p < end &&
           (isdigit((unsigned char)*p) || *p == '.' || *p == 'e' ||
            *p == 'E' || *p == '-' || *p == '+')) {
      if (*p == '.' || *p == 'e' || *p == 'E') isint = false;
      ++p;
    }
    if (p == s) {
      fail = true;
      return nullptr;
    }
    std::string t(s, p);
    if (isint) return Value::mkint(strtoll(t.c_str(), nullptr, 10));
    return Value::mknum(strtod(t.c_str(), nullptr));
  }
  ValuePtr object() {
    auto v = Value::mkobj();
    ++p; /* { */
    ws();
    if (p < end && *p == '}') {
      ++p;
      return v;
    }
    while (p < end) {
      ws();
      std::string key;
      if (!string(key)) return nullptr;
      ws();
      if (p >= end || *p != ':') {
        fail = true;
        return nullptr;
      }
      ++p;
      auto val = value();
      if (fail) return nullptr;
      v->obj[key] = val;
      ws();
      if (p < end && *p == ',') {
        ++p;
        continue;
      }
      if (p < end && *p == '}') {
        ++p;
        return v;
      }
      fail = true;
      return nullptr;
    }
    fail = true;
    return nullptr;
  }
  ValuePtr array() {
    auto v = Value::mkarr();
    ++p; /* [ */
    ws();
    if (p < end && *p == ']') {
      ++p;
      return v;
    }
    while (p < end) {
      auto val = value();
      if (fail) return nullptr;
      v->arr.push_back(val);
      ws();
      if (p < end && *p == ',') {
        ++p;
        continue;
      }
      if (p < end && *p == ']') {
        ++p;
        return v;
      }
      fail = true;
      return nullptr;
    }
    fail = true;
    return nullptr;
  }
};

void esc(const std::string &s, std::string &out) {
  out += '"';
  for (char c : s) {
    switch (c) {
      case '"': out += "\\\""; break;
      case '\\': out += "\\\\"; break;
      case '\n': out += "\\n"; break;
      case '\t': out += "\\t"; break;
      case '\r': out += "\\r"; break;
      default: out += c;
    }
  }
  out += '"';
}

void dump_rec(const ValuePtr &v, std::string &out) {
  if (!v || v->type == Value::NUL) {
    out += "null";
    return;
  }
  switch (v->type) {
    case Value::BOOL: out += v->b ? "true" : "false"; break;
    case Value::NUM: {
      char buf[64];
      if (v->is_int)
        snprintf(buf, sizeof buf, "%lld", v->inum);
      else if (std::isfinite(v->num))
        snprintf(buf, sizeof buf, "%.17g", v->num);
      else
        snprintf(buf, sizeof buf, "null"); /* JSON has no NaN/Inf */
      out += buf;
      break;
    }
    case Value::STR: esc(v->str, out); break;
    case Value::ARR: {
      out += '[';
      bool first = true;
      for (auto &e : v->arr) {
        if (!first) out += ',';
        first = false;
        dump_rec(e, out);
      }
      out += ']';
      break;
    }
    case Value::OBJ: {
      out += '{';
      bool first = true;
      for (auto &kv : v->obj) {
        if (!first) out += ',';
        first = false;
        esc(kv.first, out);
        out += ':';
        dump_rec(kv.second, out);
      }
      out += '}';
      break;
    }
    default: out += "null";
  }
}
}  // namespace

ValuePtr parse(const std::string &text) {
  Parser ps{text.data(), text.data() + text.size()};
  auto v = ps.value();
  if (ps.fail) return nullptr;
  return v;
}

std::string dump(const ValuePtr &v, int) {
  std::string out;
  dump_rec(v, out);
  return out;
}

ValuePtr get_path(const ValuePtr &root, const std::string &path) {
  ValuePtr cur = root;
  size_t pos = 0;
  while (cur && pos <= path.size()) {
    size_t slash = path.find('/', pos);
    std::string key = path.substr(pos, slash == std::string::npos
                                           ? std::string::npos
                                           : slash - pos);
    if (!cur || cur->type != Value::OBJ) return nullptr;
    auto it = cur->obj.find(key);
    if (it == cur->obj.end()) return nullptr;
    cur = it->second;
    if (slash == std::string::npos) return cur;
    pos = slash + 1;
  }
  return cur;
}

void set_path(ValuePtr root, const std::string &path, ValuePtr v) {
  ValuePtr cur = root;
  size_t pos = 0;
  while (true) {
    size_t slash = path.find('/', pos);
    std::string key = path.substr(pos, slash == std::string::npos
                                           ? std::string::npos
                                           : slash - pos);
    if (slash == std::string::npos) {
      cur->obj[key] = v;
      return;
    }
    auto it = cur->obj.find(key);
    if (it == cur->obj.end() || !it->second ||
        it->second->type != Value::OBJ)
      cur->obj[key] = Value::mkobj();
    cur = cur->obj[key];
    pos = slash + 1;
  }
}

}  // namespace bsj
