/* Minimal JSON value (parse + emit) for N5 attributes.json handling.
 * Supports the subset N5/BigStitcher metadata uses: objects, arrays,
 * strings, numbers, booleans, null. Not a general-purpose library. */
#ifndef BS_JSON_H
#define BS_JSON_H

#include <map>
#include <memory>
#include <sstream>
#include <string>
#include <vector>

namespace bsj {

struct Value;
using ValuePtr = std::shared_ptr<Value>;

struct Value {
  enum Type { NUL, BOOL, NUM, STR, ARR, OBJ } type = NUL;
  bool b = false;
  double num = 0.0;
  bool is_int = false;
  long long inum = 0;
  std::string str;
  std::vector<ValuePtr> arr;
  std::map<std::string, ValuePtr> obj;

  static ValuePtr mknull() { return std::make_shared<Value>(); }
  static ValuePtr mkbool(bool v) {
    auto p = std::make_shared<Value>();
    p->type = BOOL;
    p->b = v;
    return p;
  }
  static ValuePtr mkint(long long v) {
    auto p = std::make_shared<Value>();
    p->type = NUM;
    p->is_int = true;
    p->inum = v;
    p->num = (double)v;
    return p;
  }
  static ValuePtr mknum(double v) {
    auto p = std::make_shared<Value>();
    p->type = NUM;
    p->num = v;
    return p;
  }
  static ValuePtr mkstr(const std::string &s) {
    auto p = std::make_shared<Value>();
    p->type = STR;
    p->str = s;
    return p;
  }
  static ValuePtr mkarr() {
    auto p = std::make_shared<Value>();
    p->type = ARR;
    return p;
  }
  static ValuePtr mkobj() {
    auto p = std::make_shared<Value>();
    p->type = OBJ;
    return p;
  }
  template <typename T>
  static ValuePtr mkints(const std::vector<T> &v) {
    auto p = mkarr();
    for (auto x : v) p->arr.push_back(mkint((long long)x));
    return p;
  }
};

/* returns nullptr on malformed input */
ValuePtr parse(const std::string &text);
std::string dump(const ValuePtr &v, int indent = 0);

/* path helpers for nested objects ("a/b/c"): get returns nullptr when
 * absent; set creates intermediate objects. */
ValuePtr get_path(const ValuePtr &root, const std::string &path);
void set_path(ValuePtr root, const std::string &path, ValuePtr v);

}  // namespace bsj

#endif
