/* Minimal XML DOM for the spim_data dataset.xml schema subset (no
 * namespaces, no CDATA, no DTD — the files are machine-generated).
 * Parse + serialize with stable child order. */
#ifndef BS_XML_H
#define BS_XML_H

#include <algorithm>
#include <map>
#include <memory>
#include <string>
#include <vector>

namespace bsx {

struct Node;
using NodePtr = std::shared_ptr<Node>;

struct Node {
  std::string tag;
  std::map<std::string, std::string> attrs;
  std::string text; /* concatenated character data */
  std::vector<NodePtr> children;

  NodePtr child(const std::string &t) const {
    for (auto &c : children)
      if (c->tag == t) return c;
    return nullptr;
  }
  std::vector<NodePtr> all(const std::string &t) const {
    std::vector<NodePtr> out;
    for (auto &c : children)
      if (c->tag == t) out.push_back(c);
    return out;
  }
  NodePtr add(const std::string &t) {
    auto n = std::make_shared<Node>();
    n->tag = t;
    children.push_back(n);
    return n;
  }
  NodePtr add_text(const std::string &t, const std::string &val) {
    auto n = add(t);
    n->text = val;
    return n;
  }
  void remove_children(const std::string &t) {
    children.erase(
        std::remove_if(children.begin(), children.end(),
                       [&](const NodePtr &c) { return c->tag == t; }),
        children.end());
  }
};

NodePtr parse(const std::string &text); /* returns root element or null */
std::string serialize(const NodePtr &root);
bool load_file(const std::string &path, NodePtr *out);
bool save_file(const std::string &path, const NodePtr &root);

}  // namespace bsx

#endif
