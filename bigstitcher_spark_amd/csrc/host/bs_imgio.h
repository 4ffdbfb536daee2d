/* Input-container facade for the CLI host: a dataset XML's ImageLoader
 * is either bdv.n5 (the reference N5ImageLoader layout,
 * "setup{s}/timepoint{t}/s{l}" u16 datasets with downsamplingFactors —
 * SparkResaveN5.java:231-240 setupBdvDatasetsN5) or the OME-ZARR BDV
 * layout this repo's `resave` writes by default (the reference default
 * since SparkResaveN5.java:85 "--N5: Export as N5 (default: OMEZARR)").
 * [PIN-OMEZARR-BDV]: per-view group "setup{s}/timepoint{t}" holding
 * OME-NGFF v0.4 multiscale 5-D arrays named "0","1",... — the level
 * naming is pinned by the reference's own log lines
 * (SparkResaveN5.java:331 `useN5 ? "N5 s0" : "OME-ZARR 0"`, :347); the
 * group path and XML serialization restate the un-vendored
 * N5ApiTools.setupBdvDatasetsOMEZARR / AllenOMEZarrLoader. */
#ifndef BS_IMGIO_H
#define BS_IMGIO_H

#include <string>
#include <vector>

#include "bs_cli_util.h"
#include "bs_mip.h"
#include "bs_n5.h"
#include "bs_spimdata.h"
#include "bs_zarr.h"

namespace bsimg {

class Input {
 public:
  explicit Input(const bssd::SpimData &sd)
      : zarr_(sd.zarr_loader), n5_(sd.n5_path), zr_(sd.n5_path) {}
  Input(const std::string &path, bool zarr)
      : zarr_(zarr), n5_(path), zr_(path) {}

  bool zarr() const { return zarr_; }

  std::string dataset(int setup, int tp, int level) const {
    if (!zarr_) return bssd::SpimData::image_dataset(setup, tp, level);
    return "setup" + std::to_string(setup) + "/timepoint" +
           std::to_string(tp) + "/" + std::to_string(level);
  }

  bool read_volume_u16(int setup, int tp, int level,
                       std::vector<uint16_t> *out,
                       std::vector<long long> *dims_xyz) const {
    if (!zarr_)
      return n5_.read_volume_u16(dataset(setup, tp, level), out, dims_xyz);
    return zr_.read_volume_u16(dataset(setup, tp, level), out, dims_xyz);
  }

  std::vector<bscli::MipLevel> read_levels(int setup, int tp) const {
    if (!zarr_) return bscli::read_levels(n5_, setup, tp);
    std::vector<bscli::MipLevel> out;
    for (int l = 0;; ++l) {
      bszarr::ArrayAttrs a;
      if (!zr_.get_array_attrs(dataset(setup, tp, l), &a)) break;
      size_t nd = a.shape.size();
      if (nd < 3) break;
      bscli::MipLevel lv;
      lv.level = l;
      lv.dims = {a.shape[nd - 1], a.shape[nd - 2], a.shape[nd - 3]};
      if (l == 0) {
        lv.f[0] = lv.f[1] = lv.f[2] = 1;
      } else if (!out.empty()) {
        /* factors from the dims ratio vs level 0 (levels are
         * ceil-divided box means, so the rounded ratio is exact) */
        for (int d = 0; d < 3; ++d)
          lv.f[d] = (long long)((double)out[0].dims[d] / lv.dims[d] + 0.5);
      }
      out.push_back(lv);
    }
    return out;
  }

 private:
  bool zarr_;
  bsn5::Container n5_;
  bszarr::Container zr_;
};

}  // namespace bsimg

#endif
