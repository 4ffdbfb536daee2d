#include "bs_xml.h"

#include <algorithm>
#include <cctype>
#include <cstring>
#include <fstream>
#include <sstream>

namespace bsx {

namespace {
struct Parser {
  const char *p, *end;
  bool fail = false;

  void ws() {
    while (p < end && isspace((unsigned char)*p)) ++p;
  }
  void skip_misc() { /* decls, comments, PIs */
    while (true) {
      ws();
      if (p + 3 < end && p[0] == '<' && p[1] == '!' && p[2] == '-' &&
          p[3] == '-') {
        const char *e = strstr(p + 4, "-->");
        if (!e) {
          fail = true;
          return;
        }
        p = e + 3;
      } else if (p + 1 < end && p[0] == '<' &&
                 (p[1] == '?' || p[1] == '!')) {
        while (p < end && *p != '>') ++p;
        if (p < end) ++p;
      } else {
        return;
      }
    }
  }
  std::string name() {
    const char *s = p;
    while (p < end && (isalnum((unsigned char)*p) || *p == '_' ||
                       *p == '-' || *p == '.' || *p == ':'))
      ++p;
    return std::string(s, p);
  }
  static void unescape(std::string &s) {
    std::string out;
    out.reserve(s.size());
    for (size_t i = 0; i < s.size(); ++i) {
      if (s[i] == '&') {
        if (s.compare(i, 4, "&lt;") == 0) { out += '<'; i += 3; }
        else if (s.compare(i, 4, "&gt;") == 0) { out += '>'; i += 3; }
        else if (s.compare(i, 5, "&amp;") == 0) { out += '&'; i += 4; }
        else if (s.compare(i, 6, "&quot;") == 0) { out += '"'; i += 5; }
        else if (s.compare(i, 6, "&apos;") == 0) { out += '\''; i += 5; }
        else out += s[i];
      } else {
        out += s[i];
      }
    }
    s.swap(out);
  }
  NodePtr element() {
    skip_misc();
    if (fail || p >= end || *p != '<') {
      fail = true;
      return nullptr;
    }
    ++p;
    auto n = std::make_shared<Node>();
    n->tag = name();
    if (n->tag.empty()) {
      fail = true;
      return nullptr;
    }
    while (true) {
      ws();
      if (p >= end) {
        fail = true;
        return nullptr;
      }
      if (*p == '/') {
        ++p;
        if (p < end && *p == '>') {
          ++p;
          return n;
        }
        fail = true;
        return nullptr;
      }
      if (*p == '>') {
        ++p;
        break;
      }
      std::string an = name();
      ws();
      if (p >= end || *p != '=') {
        fail = true;
        return nullptr;
      }
      ++p;
      ws();
      if (p >= end || (*p != '"' && *p != '\'')) {
        fail = true;
        return nullptr;
      }
      char q = *p++;
      const char *s = p;
      while (p < end && *p != q) ++p;
      std::string av(s, p);
      unescape(av);
      n->attrs[an] = av;
      if (p < end) ++p;
    }
    /* content */
    while (true) {
      const char *s = p;
      while (p < end && *p != '<') ++p;
      if (p > s) {
        std::string t(s, p);
        /* keep non-whitespace character data */
        if (t.find_first_not_of(" \t\r\n") != std::string::npos) {
          size_t a = t.find_first_not_of(" \t\r\n");
          size_t b = t.find_last_not_of(" \t\r\n");
          std::string tt = t.substr(a, b - a + 1);
          unescape(tt);
          if (!n->text.empty()) n->text += " ";
          n->text += tt;
        }
      }
      if (p >= end) {
        fail = true;
        return nullptr;
      }
      if (p + 1 < end && p[1] == '/') {
        p += 2;
        std::string close = name();
        ws();
        if (p < end && *p == '>') {
          ++p;
          if (close != n->tag) fail = true;
          return n;
        }
        fail = true;
        return nullptr;
      }
      if (p + 3 < end && p[1] == '!' && p[2] == '-' && p[3] == '-') {
        const char *e = strstr(p + 4, "-->");
        if (!e) {
          fail = true;
          return nullptr;
        }
        p = e + 3;
        continue;
      }
      auto c = element();
      if (fail) return nullptr;
      n->children.push_back(c);
    }
  }
};

void escape_into(const std::string &s, std::string &out, bool attr) {
  for (char c : s) {
    switch (c) {
      case '<': out += "&lt;"; break;
      case '>': out += "&gt;"; break;
      case '&': out += "&amp;"; break;
      case '"': if (attr) { out += "&quot;"; break; }
        [[fallthrough]];
      default: out += c;
    }
  }
}

void ser(const NodePtr &n, std::string &out, int depth) {
  out.append(depth * 2, ' ');
  out += '<';
  out += n->tag;
  for (auto &kv : n->attrs) {
    out += ' ';
    out += kv.first;
    out += "=\"";
    escape_into(kv.second, out, true);
    out += '"';
  }
  if (n->children.empty() && n->text.empty()) {
    out += " />\n";
    return;
  }
  out += '>';
  if (!n->text.empty()) escape_into(n->text, out, false);
  if (!n->children.empty()) {
    out += '\n';
    for (auto &c : n->children) ser(c, out, depth + 1);
    out.append(depth * 2, ' ');
  }
  out += "</";
  out += n->tag;
  out += ">\n";
}
}  // namespace

NodePtr parse(const std::string &text) {
  Parser ps{text.data(), text.data() + text.size()};
  auto n = ps.element();
  return ps.fail ? nullptr : n;
}

std::string serialize(const NodePtr &root) {
  std::string out = "<?xml version=\"1.0\" encoding=\"UTF-8\"?>\n";
  ser(root, out, 0);
  return out;
}

bool load_file(const std::string &path, NodePtr *out) {
  std::ifstream f(path, std::ios::binary);
  if (!f) return false;
  std::string text((std::istreambuf_iterator<char>(f)),
                   std::istreambuf_iterator<char>());
  *out = parse(text);
  return *out != nullptr;
}

bool save_file(const std::string &path, const NodePtr &root) {
  std::ofstream f(path, std::ios::binary | std::ios::trunc);
  if (!f) return false;
  std::string s = serialize(root);
  f.write(s.data(), (std::streamsize)s.size());
  return (bool)f;
}

}  // namespace bsx
