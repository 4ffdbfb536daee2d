/* `create-fusion-container` — drop-in for the reference's
 * CreateFusionContainer (plain host tool, no Spark/GPU; reference
 * CreateFusionContainer.java). Creates the output N5 container with the
 * exact Bigstitcher-Spark root-attribute contract the fusion step
 * reads back (written :302-320, read SparkAffineFusion.java:239-307):
 * FusionFormat, InputXML, NumTimepoints, NumChannels, Boundingbox_min/
 * max, PreserveAnisotropy [,AnisotropyFactor], DataType, BlockSize,
 * [Min/MaxIntensity], MultiResolutionInfos. Datasets are the plain-N5
 * per-(channel,timepoint) layout "ch{c}tp{t}/s{l}" (:490-516).
 * Round-1 scope: N5 only (no ZARR/HDF5), level s0 only (the pyramid is
 * SURVEY.md §8(f) row 1), no anisotropy split. */
#include <cstdio>

#include "bs_cli_util.h"
#include "bs_n5.h"
#include "bs_zarr.h"
#include "bs_spimdata.h"

int main(int argc, char **argv) {
  bscli::Args args;
  std::map<std::string, std::string> alias = {{"-x", "--xml"},
                                              {"-o", "--n5Path"}};
  if (!args.parse(argc, argv, alias, {"preserveAnisotropy"}) ||
      !args.has("xml") || !args.has("n5Path")) {
    fprintf(stderr,
            "usage: create-fusion-container -x dataset.xml -o out.n5 "
            "[--blockSize 128,128,128] [--dataType UINT16|UINT8|FLOAT32] "
            "[--minIntensity v --maxIntensity v] [--bbMin x,y,z --bbMax "
            "x,y,z] [--compression GZIP|RAW] [--storage N5|ZARR] "
            "[--downsamplings \"1,1,1;2,2,2;4,4,4\"]\n");
    return 2;
  }
  bssd::SpimData sd;
  std::string err;
  if (!sd.load(args.get("xml"), &err)) {
    fprintf(stderr, "error: %s\n", err.c_str());
    return 1;
  }
  auto bs = bscli::parse_ints(args.get("blockSize", "128,128,128"));
  std::string dt = args.get("dataType", "UINT16");
  std::string n5dt = dt == "UINT8" ? "uint8"
                     : dt == "FLOAT32" ? "float32" : "uint16";
  std::string comp =
      args.get("compression", "GZIP") == "RAW" ? "raw" : "gzip";

  /* bounding box: explicit or the union of transformed view bboxes
   * (the reference's default "estimate bounding box", :122-211) */
  long long bbmin[3], bbmax[3];
  if (args.has("bbMin") && args.has("bbMax")) {
    auto mn = bscli::parse_ints(args.get("bbMin"));
    auto mx = bscli::parse_ints(args.get("bbMax"));
    for (int d = 0; d < 3; ++d) {
      bbmin[d] = mn[d];
      bbmax[d] = mx[d];
    }
  } else {
    double lo[3] = {1e300, 1e300, 1e300}, hi[3] = {-1e300, -1e300, -1e300};
    int tp0 = sd.timepoints.empty() ? 0 : sd.timepoints[0];
    for (auto &s : sd.setups) {
      auto r = sd.regs.find({tp0, s.id});
      if (r == sd.regs.end()) continue;
      double l[3], h[3];
      bscli::tbbox(r->second, s.dims, l, h);
      for (int d = 0; d < 3; ++d) {
        lo[d] = std::min(lo[d], l[d]);
        hi[d] = std::max(hi[d], h[d]);
      }
    }
    for (int d = 0; d < 3; ++d) {
      bbmin[d] = (long long)std::floor(lo[d]);
      bbmax[d] = (long long)std::ceil(hi[d]);
    }
  }
  long long dims[3] = {bbmax[0] - bbmin[0] + 1, bbmax[1] - bbmin[1] + 1,
                       bbmax[2] - bbmin[2] + 1};
  int numTp = (int)sd.timepoints.size(), numCh = 1;
  const bool zarr = args.get("storage", "N5") == "ZARR";

  bsn5::Container n5(args.get("n5Path"));
  bszarr::Container zr(args.get("n5Path"));
  if (zarr ? !zr.create() : !n5.create()) {
    fprintf(stderr, "cannot create %s\n", args.get("n5Path").c_str());
    return 1;
  }
  auto set = [&](const std::string &k, bsj::ValuePtr v) {
    if (zarr)
      zr.set_root_attr("Bigstitcher-Spark/" + k, v);
    else
      n5.set_attr("", "Bigstitcher-Spark/" + k, v);
  };
  set("FusionFormat", bsj::Value::mkstr(zarr ? "OME-ZARR" : "N5"));
  set("InputXML", bsj::Value::mkstr(args.get("xml")));
  set("NumTimepoints", bsj::Value::mkint(numTp));
  set("NumChannels", bsj::Value::mkint(numCh));
  set("Boundingbox_min",
      bsj::Value::mkints(std::vector<long long>{bbmin[0], bbmin[1], bbmin[2]}));
  set("Boundingbox_max",
      bsj::Value::mkints(std::vector<long long>{bbmax[0], bbmax[1], bbmax[2]}));
  set("PreserveAnisotropy", bsj::Value::mkbool(false));
  set("DataType", bsj::Value::mkstr(dt));
  set("BlockSize", bsj::Value::mkints(std::vector<long long>{bs[0], bs[1], bs[2]}));
  if (args.has("minIntensity") && args.has("maxIntensity")) {
    set("MinIntensity", bsj::Value::mknum(args.getd("minIntensity", 0)));
    set("MaxIntensity", bsj::Value::mknum(args.getd("maxIntensity", 65535)));
  }
  /* pyramid ladder (reference
   * CreateFusionContainer.java:260-273 estimates it — here explicit via
   * --downsamplings, default s0 only) */
  std::vector<std::array<long long, 3>> ladder;
  {
    std::string spec = args.get("downsamplings", "1,1,1");
    std::string cur;
    for (char ch2 : spec + ";") {
      if (ch2 == ';') {
        auto f = bscli::parse_ints(cur);
        if (f.size() == 3) ladder.push_back({f[0], f[1], f[2]});
        cur.clear();
      } else {
        cur += ch2;
      }
    }
    if (ladder.empty() || ladder[0][0] != 1 || ladder[0][1] != 1 ||
        ladder[0][2] != 1) {
      fprintf(stderr, "--downsamplings must start with 1,1,1\n");
      return 2;
    }
  }
  auto mri_all = bsj::Value::mkarr();
  if (zarr) {
    /* ONE 5-D [t,c,z,y,x] array per level, OME-NGFF v0.4 multiscales
     * (reference CreateFusionContainer.java:331-389) */
    std::string zdt = n5dt == "uint8" ? "|u1"
                      : n5dt == "float32" ? "<f4" : "<u2";
    auto msets = bsj::Value::mkarr();
    for (size_t l = 0; l < ladder.size(); ++l) {
      char dsname[32];
      snprintf(dsname, sizeof dsname, "s%zu", l);
      bszarr::ArrayAttrs za;
      za.shape = {numTp, numCh,
                  (dims[2] + ladder[l][2] - 1) / ladder[l][2],
                  (dims[1] + ladder[l][1] - 1) / ladder[l][1],
                  (dims[0] + ladder[l][0] - 1) / ladder[l][0]};
      za.chunks = {1, 1, (int)bs[2], (int)bs[1], (int)bs[0]};
      za.dtype = zdt;
      za.gzip = comp == "gzip";
      if (!zr.create_array(dsname, za)) {
        fprintf(stderr, "cannot create array %s\n", dsname);
        return 1;
      }
      auto dset = bsj::Value::mkobj();
      dset->obj["path"] = bsj::Value::mkstr(dsname);
      auto cts = bsj::Value::mkarr();
      auto sc = bsj::Value::mkobj();
      sc->obj["type"] = bsj::Value::mkstr("scale");
      auto scale = bsj::Value::mkarr();
      scale->arr.push_back(bsj::Value::mknum(1.0));
      scale->arr.push_back(bsj::Value::mknum(1.0));
      scale->arr.push_back(bsj::Value::mknum((double)ladder[l][2]));
      scale->arr.push_back(bsj::Value::mknum((double)ladder[l][1]));
      scale->arr.push_back(bsj::Value::mknum((double)ladder[l][0]));
      sc->obj["scale"] = scale;
      cts->arr.push_back(sc);
      dset->obj["coordinateTransformations"] = cts;
      msets->arr.push_back(dset);
      /* one shared MultiResolutionInfos entry set (all t,c share it) */
      if (l == 0)
        for (int vi = 0; vi < numTp * numCh; ++vi)
          mri_all->arr.push_back(bsj::Value::mkarr());
      for (int vi = 0; vi < numTp * numCh; ++vi) {
        auto lv = bsj::Value::mkobj();
        lv->obj["dataset"] = bsj::Value::mkstr(dsname);
        lv->obj["dimensions"] = bsj::Value::mkints(
            std::vector<long long>{za.shape[4], za.shape[3], za.shape[2]});
        lv->obj["blockSize"] = bsj::Value::mkints(
            std::vector<int>{(int)bs[0], (int)bs[1], (int)bs[2]});
        lv->obj["absoluteDownsampling"] = bsj::Value::mkints(
            std::vector<long long>{ladder[l][0], ladder[l][1],
                                   ladder[l][2]});
        mri_all->arr[vi]->arr.push_back(lv);
      }
    }
    auto ms = bsj::Value::mkobj();
    ms->obj["version"] = bsj::Value::mkstr("0.4");
    ms->obj["name"] = bsj::Value::mkstr("fused");
    auto axes = bsj::Value::mkarr();
    const char *axn[5] = {"t", "c", "z", "y", "x"};
    const char *axt[5] = {"time", "channel", "space", "space", "space"};
    for (int i = 0; i < 5; ++i) {
      auto ax = bsj::Value::mkobj();
      ax->obj["name"] = bsj::Value::mkstr(axn[i]);
      ax->obj["type"] = bsj::Value::mkstr(axt[i]);
      axes->arr.push_back(ax);
    }
    ms->obj["axes"] = axes;
    ms->obj["datasets"] = msets;
    auto msl = bsj::Value::mkarr();
    msl->arr.push_back(ms);
    zr.set_root_attr("multiscales", msl);
  } else {
    for (int t = 0; t < numTp; ++t)
      for (int ch = 0; ch < numCh; ++ch) {
        auto levels = bsj::Value::mkarr();
        for (size_t l = 0; l < ladder.size(); ++l) {
          char dsname[64];
          snprintf(dsname, sizeof dsname, "ch%dtp%d/s%zu", ch, t, l);
          bsn5::DatasetAttrs da;
          da.dims = {(dims[0] + ladder[l][0] - 1) / ladder[l][0],
                     (dims[1] + ladder[l][1] - 1) / ladder[l][1],
                     (dims[2] + ladder[l][2] - 1) / ladder[l][2]};
          da.block = {(int)bs[0], (int)bs[1], (int)bs[2]};
          da.dtype = n5dt;
          da.compression = comp;
          if (!n5.create_dataset(dsname, da)) {
            fprintf(stderr, "cannot create dataset %s\n", dsname);
            return 1;
          }
          auto lv = bsj::Value::mkobj();
          lv->obj["dataset"] = bsj::Value::mkstr(dsname);
          lv->obj["dimensions"] = bsj::Value::mkints(da.dims);
          lv->obj["blockSize"] =
              bsj::Value::mkints(std::vector<int>{da.block[0], da.block[1],
                                                  da.block[2]});
          lv->obj["absoluteDownsampling"] = bsj::Value::mkints(
              std::vector<long long>{ladder[l][0], ladder[l][1],
                                     ladder[l][2]});
          levels->arr.push_back(lv);
        }
        mri_all->arr.push_back(levels);
      }
  }
  set("MultiResolutionInfos", mri_all);
  printf("created %s: %d tp x %d ch, bbox [%lld,%lld,%lld]..[%lld,%lld,%lld]"
         ", %s %s\n",
         args.get("n5Path").c_str(), numTp, numCh, bbmin[0], bbmin[1],
         bbmin[2], bbmax[0], bbmax[1], bbmax[2], dt.c_str(), comp.c_str());
  return 0;
}
