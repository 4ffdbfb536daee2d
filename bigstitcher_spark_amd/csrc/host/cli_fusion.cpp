/* `affine-fusion` — drop-in for the reference's SparkAffineFusion CLI
 * (reference SparkAffineFusion.java). Reads the container's
 * Bigstitcher-Spark metadata contract (:239-307), decomposes the
 * bounding box into the output grid (Grid.create, :459-461), culls views
 * per block (OverlappingViews, :536-537), fuses every block on the GPU
 * (bs_fuse_blocks = the BlkAffineFusion replacement, :602-615) and
 * writes the voxel blocks into the container (N5Utils.saveBlock, :670).
 * Round-1 scope: plain-N5 containers, level s0, FusionTypes
 * AVG/AVG_BLEND/MAX_INTENSITY, no intensity coefficients / masks. */
#include <cstdio>
#include <set>

#include "../../../include/bigstitch.h"
#include "bs_cli_util.h"
#include "bs_n5.h"
#include "bs_spimdata.h"

int main(int argc, char **argv) {
  bscli::Args args;
  std::map<std::string, std::string> alias = {{"-x", "--xml"},
                                              {"-o", "--n5Path"}};
  if (!args.parse(argc, argv, alias, {"masks"}) || !args.has("n5Path")) {
    fprintf(stderr,
            "usage: affine-fusion -o out.n5 [-x dataset.xml] "
            "[--fusionType AVG_BLEND|AVG|MAX_INTENSITY] "
            "[--blendingRange 40] [--blendingBorder 0] [--device N] "
            "[--batchSize 64]\n");
    return 2;
  }
  bsn5::Container n5(args.get("n5Path"));
  auto geta = [&](const std::string &k) {
    return n5.get_attr("", "Bigstitcher-Spark/" + k);
  };
  auto fmt = geta("FusionFormat");
  if (!fmt) {
    fprintf(stderr,
            "Could not load 'Bigstitcher-Spark/FusionFormat' from '%s'. "
            "Note: this metadata is created by create-fusion-container "
            "in the previous step.\n",
            args.get("n5Path").c_str());
    return 1;
  }
  std::string xml = args.get("xml");
  if (xml.empty()) {
    auto x = geta("InputXML");
    if (x) xml = x->str;
  }
  auto bbmin_a = geta("Boundingbox_min"), bbmax_a = geta("Boundingbox_max");
  auto bs_a = geta("BlockSize");
  auto dt_a = geta("DataType");
  auto ntp_a = geta("NumTimepoints");
  if (!bbmin_a || !bbmax_a || !bs_a || !dt_a) {
    fprintf(stderr, "incomplete container metadata\n");
    return 1;
  }
  long long bbmin[3], bbmax[3];
  int blk[3];
  for (int d = 0; d < 3; ++d) {
    bbmin[d] = bbmin_a->arr[d]->inum;
    bbmax[d] = bbmax_a->arr[d]->inum;
    blk[d] = (int)bs_a->arr[d]->inum;
  }
  long long dims[3] = {bbmax[0] - bbmin[0] + 1, bbmax[1] - bbmin[1] + 1,
                       bbmax[2] - bbmin[2] + 1};
  std::string dt = dt_a->str;
  int out_dtype = dt == "UINT8" ? BS_OUT_UINT8
                  : dt == "FLOAT32" ? BS_OUT_FLOAT32 : BS_OUT_UINT16;
  double minI = 0, maxI = 65535;
  auto minI_a = geta("MinIntensity"), maxI_a = geta("MaxIntensity");
  if (minI_a) minI = minI_a->is_int ? minI_a->inum : minI_a->num;
  if (maxI_a) maxI = maxI_a->is_int ? maxI_a->inum : maxI_a->num;
  if (out_dtype == BS_OUT_UINT8 && !maxI_a) maxI = 255;

  std::string ft = args.get("fusionType", "AVG_BLEND");
  int fusion_type = ft == "AVG" ? BS_FUSION_AVG
                    : ft == "MAX_INTENSITY" ? BS_FUSION_MAX_INTENSITY
                                            : BS_FUSION_AVG_BLEND;
  float brange = (float)args.getd("blendingRange", 40.0);
  float bborder = (float)args.getd("blendingBorder", 0.0);

  bssd::SpimData sd;
  std::string err;
  if (!sd.load(xml, &err)) {
    fprintf(stderr, "error loading xml '%s': %s\n", xml.c_str(),
            err.c_str());
    return 1;
  }
  bsn5::Container in_n5(sd.n5_path);
  bs_ctx *ctx = nullptr;
  if (bs_ctx_create(&ctx, (int)args.getl("device", 0)) != BS_OK) {
    fprintf(stderr, "error: %s\n", bs_last_error(nullptr));
    return 1;
  }
  int num_tp = ntp_a ? (int)ntp_a->inum : (int)sd.timepoints.size();
  size_t batch = (size_t)args.getl("batchSize", 64);

  for (int ti = 0; ti < num_tp; ++ti) {
    int tp = sd.timepoints[ti % sd.timepoints.size()];
    /* upload + describe views of this timepoint */
    std::vector<bs_fuse_view> fviews;
    std::vector<const bssd::ViewSetup *> fsetups;
    for (auto &s : sd.setups) {
      auto r = sd.regs.find({tp, s.id});
      if (r == sd.regs.end()) continue;
      std::vector<uint16_t> vox;
      std::vector<long long> vdims;
      if (!in_n5.read_volume_u16(bssd::SpimData::image_dataset(s.id, tp),
                                 &vox, &vdims)) {
        fprintf(stderr, "cannot read view tp=%d setup=%d\n", tp, s.id);
        return 1;
      }
      int64_t d[3] = {vdims[0], vdims[1], vdims[2]};
      if (bs_view_upload(ctx, s.id, vox.data(), d) != BS_OK) {
        fprintf(stderr, "upload failed: %s\n", bs_last_error(ctx));
        return 1;
      }
      bs_fuse_view fv{};
      fv.view_id = s.id;
      /* world = bbox coords: shift model by -bbmin */
      for (int i = 0; i < 12; ++i) fv.affine[i] = r->second[i];
      fv.affine[3] -= bbmin[0];
      fv.affine[7] -= bbmin[1];
      fv.affine[11] -= bbmin[2];
      for (int d2 = 0; d2 < 3; ++d2) {
        fv.blend_border[d2] = bborder;
        fv.blend_range[d2] = brange;
      }
      fviews.push_back(fv);
      fsetups.push_back(&s);
    }
    char dsname[64];
    snprintf(dsname, sizeof dsname, "ch0tp%d/s0", ti);
    bsn5::DatasetAttrs da;
    if (!n5.get_dataset_attrs(dsname, &da)) {
      fprintf(stderr, "missing dataset %s in container\n", dsname);
      return 1;
    }
    /* output grid (Grid.create semantics) + per-block culling */
    size_t esz = bsn5::dtype_size(da.dtype);
    long long ngx = (dims[0] + blk[0] - 1) / blk[0];
    long long ngy = (dims[1] + blk[1] - 1) / blk[1];
    long long ngz = (dims[2] + blk[2] - 1) / blk[2];
    std::vector<bs_block_desc> blocks;
    std::vector<std::vector<long long>> gps;
    std::vector<int32_t> vidx;
    std::vector<int64_t> voffs;
    std::vector<std::vector<char>> outbufs;
    std::vector<void *> outptrs;
    long long nblocks_total = ngx * ngy * ngz, done = 0;
    auto flush = [&]() -> bool {
      if (blocks.empty()) return true;
      voffs.push_back((int64_t)vidx.size());
      bs_fuse_params prm{};
      prm.fusion_type = fusion_type;
      prm.out_dtype = out_dtype;
      prm.min_intensity = minI;
      prm.max_intensity = maxI;
      prm.interp = 1;
      if (bs_fuse_blocks(ctx, fviews.data(), fviews.size(), blocks.data(),
                         blocks.size(), vidx.data(), voffs.data(), &prm,
                         outptrs.data()) != BS_OK) {
        fprintf(stderr, "fusion failed: %s\n", bs_last_error(ctx));
        return false;
      }
      for (size_t b = 0; b < blocks.size(); ++b) {
        std::vector<int> clipped = {(int)blocks[b].size[0],
                                    (int)blocks[b].size[1],
                                    (int)blocks[b].size[2]};
        if (!n5.write_block(dsname, da, gps[b], outbufs[b].data(),
                            clipped)) {
          fprintf(stderr, "block write failed\n");
          return false;
        }
      }
      done += (long long)blocks.size();
      printf("fused %lld / %lld blocks\n", done, nblocks_total);
      blocks.clear();
      gps.clear();
      vidx.clear();
      voffs.clear();
      outbufs.clear();
      outptrs.clear();
      return true;
    };
    for (long long gz = 0; gz < ngz; ++gz)
      for (long long gy = 0; gy < ngy; ++gy)
        for (long long gx = 0; gx < ngx; ++gx) {
          bs_block_desc bd{};
          bd.min[0] = gx * blk[0];
          bd.min[1] = gy * blk[1];
          bd.min[2] = gz * blk[2];
          bd.size[0] = std::min((long long)blk[0], dims[0] - bd.min[0]);
          bd.size[1] = std::min((long long)blk[1], dims[1] - bd.min[1]);
          bd.size[2] = std::min((long long)blk[2], dims[2] - bd.min[2]);
          /* cull: transformed view bbox (+2 px guard) vs block
           * (OverlappingViews.findOverlappingViews, ref :28-47) */
          if (blocks.empty()) voffs.push_back(0);
          else voffs.push_back((int64_t)vidx.size());
          for (size_t v = 0; v < fviews.size(); ++v) {
            double lo[3], hi[3];
            bscli::M34 m;
            for (int i = 0; i < 12; ++i) m[i] = fviews[v].affine[i];
            bscli::tbbox(m, fsetups[v]->dims, lo, hi);
            bool ov = true;
            for (int d = 0; d < 3; ++d)
              if (hi[d] + 2 < bd.min[d] ||
                  lo[d] - 2 > bd.min[d] + bd.size[d])
                ov = false;
            if (ov) vidx.push_back((int32_t)v);
          }
          blocks.push_back(bd);
          gps.push_back({gx, gy, gz});
          outbufs.emplace_back((size_t)bd.size[0] * bd.size[1] *
                               bd.size[2] * esz);
          outptrs.push_back(outbufs.back().data());
          if (blocks.size() >= batch) {
            if (!flush()) return 1;
          }
        }
    if (!flush()) return 1;
    for (auto &fv : fviews) bs_view_release(ctx, fv.view_id);
  }
  bs_ctx_destroy(ctx);
  printf("affine-fusion done.\n");
  return 0;
}
