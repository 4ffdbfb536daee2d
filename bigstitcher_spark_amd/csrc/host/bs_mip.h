/* Multi-resolution input helpers: enumerate a view's pyramid levels in
 * a bdv.n5 container ("setup{s}/timepoint{t}/s{l}" datasets with a
 * "downsamplingFactors" attribute per level, as written by `resave`)
 * and the level-pick rules (bs_cli_util.h pick_level_for_transform,
 * restating ViewUtil.java:425-493). */
#ifndef BS_MIP_H
#define BS_MIP_H

#include "bs_cli_util.h"
#include "bs_n5.h"
#include "bs_spimdata.h"

namespace bscli {

inline std::vector<MipLevel> read_levels(const bsn5::Container &n5,
                                         int setup, int tp) {
  std::vector<MipLevel> out;
  for (int l = 0;; ++l) {
    std::string name = bssd::SpimData::image_dataset(setup, tp, l);
    bsn5::DatasetAttrs a;
    if (!n5.get_dataset_attrs(name, &a)) break;
    MipLevel lv;
    lv.level = l;
    lv.dims = a.dims;
    auto df = n5.get_attr(name, "downsamplingFactors");
    if (df && df->type == bsj::Value::ARR && df->arr.size() == 3)
      for (int d = 0; d < 3; ++d) lv.f[d] = df->arr[d]->inum;
    out.push_back(lv);
  }
  return out;
}

/* [PIN-MIPSEL] restatement of mvrecon DownsampleTools.openAndDownsample
 * for the stitching path (called behind computeStitching's ds
 * parameter; artifact un-vendored): open the existing level with the
 * LARGEST factors that divide the requested downsampling on every
 * axis, then box-downsample the remainder. Returns the level index
 * (always valid: level 0 has factors 1,1,1). */
inline int pick_level_dividing(const std::vector<MipLevel> &levels,
                               const long long ds[3]) {
  int best = 0;
  long long bestProd = 1;
  for (size_t l = 0; l < levels.size(); ++l) {
    const auto &lv = levels[l];
    bool ok = true;
    for (int d = 0; d < 3; ++d)
      if (lv.f[d] <= 0 || ds[d] % lv.f[d] != 0) ok = false;
    long long prod = lv.f[0] * lv.f[1] * lv.f[2];
    if (ok && prod > bestProd) {
      bestProd = prod;
      best = (int)l;
    }
  }
  return best;
}

}  // namespace bscli

#endif
