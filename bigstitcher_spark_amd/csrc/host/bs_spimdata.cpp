#include "bs_spimdata.h"

#include <algorithm>
#include <cmath>
#include <cstdio>
#include <cstring>
#include <sstream>

namespace bssd {

namespace {

std::vector<double> nums(const std::string &s) {
  std::vector<double> out;
  std::istringstream is(s);
  double v;
  while (is >> v) out.push_back(v);
  return out;
}

std::string join(const double *v, int n) {
  std::ostringstream os;
  os.precision(17);
  for (int i = 0; i < n; ++i) {
    if (i) os << " ";
    os << v[i];
  }
  return os.str();
}

/* c = a * b for row-major 3x4 affines (homogeneous last row 0 0 0 1) */
std::array<double, 12> matmul(const std::array<double, 12> &a,
                              const std::array<double, 12> &b) {
  std::array<double, 12> c{};
  for (int r = 0; r < 3; ++r) {
    for (int k = 0; k < 4; ++k) {
      double s = 0;
      for (int j = 0; j < 3; ++j) s += a[r * 4 + j] * b[j * 4 + k];
      if (k == 3) s += a[r * 4 + 3];
      c[r * 4 + k] = s;
    }
  }
  return c;
}

std::string dirname_of(const std::string &p) {
  size_t s = p.find_last_of('/');
  return s == std::string::npos ? std::string(".") : p.substr(0, s);
}

std::string viewids_str(const std::vector<ViewId> &v) {
  std::ostringstream os;
  for (size_t i = 0; i < v.size(); ++i) {
    if (i) os << ";";
    os << v[i].first << "," << v[i].second;
  }
  return os.str();
}

std::vector<ViewId> viewids_parse(const std::string &s) {
  std::vector<ViewId> out;
  std::istringstream is(s);
  std::string tok;
  while (std::getline(is, tok, ';')) {
    int tp, su;
    if (sscanf(tok.c_str(), "%d,%d", &tp, &su) == 2)
      out.push_back({tp, su});
  }
  return out;
}

}  // namespace

bool SpimData::load(const std::string &path, std::string *err) {
  xml_path = path;
  base_dir = dirname_of(path);
  if (!bsx::load_file(path, &root)) {
    if (err) *err = "cannot parse " + path;
    return false;
  }
  auto seq = root->child("SequenceDescription");
  if (!seq) {
    if (err) *err = "missing SequenceDescription";
    return false;
  }
  auto loader = seq->child("ImageLoader");
  if (loader) {
    auto n5 = loader->child("n5");
    auto zr = loader->child("zarr"); /* [PIN-OMEZARR-BDV] bs_imgio.h */
    auto src = n5 ? n5 : zr;
    if (src) {
      bool rel =
          !src->attrs.count("type") || src->attrs["type"] == "relative";
      n5_path = rel ? base_dir + "/" + src->text : src->text;
      zarr_loader = (src == zr);
    }
  }
  auto vss = seq->child("ViewSetups");
  if (!vss) {
    if (err) *err = "missing ViewSetups";
    return false;
  }
  setups.clear();
  for (auto &vs : vss->all("ViewSetup")) {
    ViewSetup s;
    auto idn = vs->child("id");
    if (!idn) continue;
    s.id = atoi(idn->text.c_str());
    auto nm = vs->child("name");
    if (nm) s.name = nm->text;
    auto sz = vs->child("size");
    if (sz) {
      auto d = nums(sz->text);
      for (size_t i = 0; i < 3 && i < d.size(); ++i)
        s.dims[i] = (long long)d[i];
    }
    auto vx = vs->child("voxelSize");
    if (vx) {
      auto vsz = vx->child("size");
      if (vsz) {
        auto d = nums(vsz->text);
        for (size_t i = 0; i < 3 && i < d.size(); ++i) s.voxel[i] = d[i];
      }
    }
    auto at = vs->child("attributes");
    if (at) {
      auto rd = [&](const char *tag, int *dst) {
        auto n = at->child(tag);
        if (n) *dst = atoi(n->text.c_str());
      };
      rd("angle", &s.angle);
      rd("tile", &s.tile);
      rd("channel", &s.channel);
      rd("illumination", &s.illumination);
    }
    setups.push_back(s);
  }
  timepoints.clear();
  auto tp = seq->child("Timepoints");
  if (tp) {
    auto pat = tp->child("integerpattern");
    if (pat) {
      std::istringstream is(pat->text);
      std::string tok;
      while (std::getline(is, tok, ','))
        timepoints.push_back(atoi(tok.c_str()));
    } else {
      auto first = tp->child("first");
      auto last = tp->child("last");
      if (first && last)
        for (int t = atoi(first->text.c_str());
             t <= atoi(last->text.c_str()); ++t)
          timepoints.push_back(t);
    }
  }
  if (timepoints.empty()) timepoints.push_back(0);
  regs.clear();
  auto vrs = root->child("ViewRegistrations");
  if (vrs) {
    for (auto &vr : vrs->all("ViewRegistration")) {
      int t = atoi(vr->attrs["timepoint"].c_str());
      int s = atoi(vr->attrs["setup"].c_str());
      std::array<double, 12> model = {1, 0, 0, 0, 0, 1, 0, 0, 0, 0, 1, 0};
      bool first = true;
      /* list order: first element is the OUTERMOST transform */
      for (auto &vt : vr->all("ViewTransform")) {
        auto aff = vt->child("affine");
        if (!aff) continue;
        auto d = nums(aff->text);
        if (d.size() != 12) continue;
        std::array<double, 12> m;
        for (int i = 0; i < 12; ++i) m[i] = d[i];
        model = first ? m : matmul(model, m);
        first = false;
      }
      regs[{t, s}] = model;
    }
  }
  return true;
}

bool SpimData::save(const std::string &path) const {
  return bsx::save_file(path, root);
}

const ViewSetup *SpimData::setup(int id) const {
  for (auto &s : setups)
    if (s.id == id) return &s;
  return nullptr;
}

std::string SpimData::image_dataset(int setup, int tp, int level) {
  char buf[96];
  snprintf(buf, sizeof buf, "setup%d/timepoint%d/s%d", setup, tp, level);
  return buf;
}

double SpimData::calculate_hash(const std::array<double, 12> &a,
                                const std::array<double, 12> &b) {
  double h = 0;
  for (double v : a) h += v;
  for (double v : b) h += v;
  return h; /* [PIN-HASH] — see header */
}

std::vector<StitchEntry> SpimData::stitching_results() const {
  std::vector<StitchEntry> out;
  auto sr = root->child("StitchingResults");
  if (!sr) return out;
  for (auto &pr : sr->all("PairwiseResult")) {
    StitchEntry e;
    auto va = pr->child("ViewIdsA"), vb = pr->child("ViewIdsB");
    if (va) e.views_a = viewids_parse(va->text);
    if (vb) e.views_b = viewids_parse(vb->text);
    auto m = pr->child("Matrix");
    if (m) {
      auto d = nums(m->text);
      for (size_t i = 0; i < 12 && i < d.size(); ++i) e.matrix[i] = d[i];
    }
    auto mn = pr->child("BoundingBoxMin"), mx = pr->child("BoundingBoxMax");
    if (mn) {
      auto d = nums(mn->text);
      for (size_t i = 0; i < 3 && i < d.size(); ++i) e.bbox_min[i] = d[i];
    }
    if (mx) {
      auto d = nums(mx->text);
      for (size_t i = 0; i < 3 && i < d.size(); ++i) e.bbox_max[i] = d[i];
    }
    auto r = pr->child("Correlation");
    if (r) e.r = atof(r->text.c_str());
    auto h = pr->child("Hash");
    if (h) e.hash = atof(h->text.c_str());
    out.push_back(e);
  }
  return out;
}

void SpimData::set_stitching_results(
    const std::vector<StitchEntry> &entries) {
  root->remove_children("StitchingResults");
  auto sr = root->add("StitchingResults");
  for (auto &e : entries) {
    auto pr = sr->add("PairwiseResult");
    pr->add_text("ViewIdsA", viewids_str(e.views_a));
    pr->add_text("ViewIdsB", viewids_str(e.views_b));
    pr->add_text("Matrix", join(e.matrix, 12));
    pr->add_text("BoundingBoxMin", join(e.bbox_min, 3));
    pr->add_text("BoundingBoxMax", join(e.bbox_max, 3));
    pr->add_text("Correlation", join(&e.r, 1));
    pr->add_text("Hash", join(&e.hash, 1));
  }
}

namespace {
/* "0,1,2" -> set; empty string -> nullopt-like empty set + false */
bool parse_idlist(const std::string &s, std::vector<int> *out) {
  if (s.empty()) return false;
  std::istringstream is(s);
  std::string tok;
  while (std::getline(is, tok, ','))
    if (!tok.empty()) out->push_back(atoi(tok.c_str()));
  return true;
}
bool contains(const std::vector<int> &v, int x) {
  for (int e : v)
    if (e == x) return true;
  return false;
}
}  // namespace

bool select_views(const SpimData &sd, const std::vector<std::string> &vi,
                  const std::string &angle_ids, const std::string &tile_ids,
                  const std::string &illum_ids,
                  const std::string &channel_ids,
                  const std::string &timepoint_ids,
                  std::vector<ViewId> *out, std::string *err) {
  out->clear();
  if (!vi.empty()) {
    /* explicit -vi 'tp,setup' list wins (Import.java:103-139) */
    for (auto &s : vi) {
      int tp, su;
      if (sscanf(s.c_str(), "%d,%d", &tp, &su) != 2) {
        if (err) *err = "bad -vi entry '" + s + "' (expected 'tp,setup')";
        return false;
      }
      if (!sd.setup(su) || !sd.regs.count({tp, su})) {
        if (err)
          *err = "-vi view (" + std::to_string(tp) + "," +
                 std::to_string(su) + ") not present in the XML";
        return false;
      }
      out->push_back({tp, su});
    }
    return true;
  }
  std::vector<int> ang, til, ill, cha, tps;
  bool f_ang = parse_idlist(angle_ids, &ang);
  bool f_til = parse_idlist(tile_ids, &til);
  bool f_ill = parse_idlist(illum_ids, &ill);
  bool f_cha = parse_idlist(channel_ids, &cha);
  bool f_tps = parse_idlist(timepoint_ids, &tps);
  for (int tp : sd.timepoints) {
    if (f_tps && !contains(tps, tp)) continue;
    for (auto &s : sd.setups) {
      if (!sd.regs.count({tp, s.id})) continue;
      if (f_ang && !contains(ang, s.angle)) continue;
      if (f_til && !contains(til, s.tile)) continue;
      if (f_ill && !contains(ill, s.illumination)) continue;
      if (f_cha && !contains(cha, s.channel)) continue;
      out->push_back({tp, s.id});
    }
  }
  if (out->empty() && err) *err = "no views selected";
  return !out->empty();
}

double average_anisotropy(const SpimData &sd,
                          const std::vector<ViewId> &views) {
  double sum = 0;
  int n = 0;
  for (auto &v : views) {
    const ViewSetup *s = sd.setup(v.second);
    if (!s) continue;
    double xy = std::min(s->voxel[0], s->voxel[1]);
    if (xy <= 0) continue;
    sum += s->voxel[2] / xy;
    ++n;
  }
  return n ? sum / n : 1.0;
}

bsx::NodePtr make_dataset_xml(
    const std::string &n5_rel, const std::vector<ViewSetup> &setups,
    const std::vector<int> &tps,
    const std::map<ViewId, std::array<double, 12>> &regs) {
  auto root = std::make_shared<bsx::Node>();
  root->tag = "SpimData";
  root->attrs["version"] = "0.2";
  auto bp = root->add_text("BasePath", ".");
  bp->attrs["type"] = "relative";
  auto seq = root->add("SequenceDescription");
  auto il = seq->add("ImageLoader");
  il->attrs["format"] = "bdv.n5";
  il->attrs["version"] = "1.0";
  auto n5 = il->add_text("n5", n5_rel);
  n5->attrs["type"] = "relative";
  auto vss = seq->add("ViewSetups");
  for (auto &s : setups) {
    auto vs = vss->add("ViewSetup");
    vs->add_text("id", std::to_string(s.id));
    vs->add_text("name", s.name.empty() ? std::to_string(s.id) : s.name);
    char buf[96];
    snprintf(buf, sizeof buf, "%lld %lld %lld", s.dims[0], s.dims[1],
             s.dims[2]);
    vs->add_text("size", buf);
  }
  auto tp = seq->add("Timepoints");
  tp->attrs["type"] = "pattern";
  std::string pat;
  for (size_t i = 0; i < tps.size(); ++i) {
    if (i) pat += ",";
    pat += std::to_string(tps[i]);
  }
  tp->add_text("integerpattern", pat);
  auto vrs = root->add("ViewRegistrations");
  for (auto &kv : regs) {
    auto vr = vrs->add("ViewRegistration");
    vr->attrs["timepoint"] = std::to_string(kv.first.first);
    vr->attrs["setup"] = std::to_string(kv.first.second);
    auto vt = vr->add("ViewTransform");
    vt->attrs["type"] = "affine";
    vt->add_text("Name", "Translation to Regular Grid");
    vt->add_text("affine", join(kv.second.data(), 12));
  }
  return root;
}

}  // namespace bssd
