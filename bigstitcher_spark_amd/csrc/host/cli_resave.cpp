/* `resave` — drop-in for the reference's SparkResaveN5 (reference
 * SparkResaveN5.java:189-207, 298-383: per-view s0 re-blocking via
 * N5ApiTools.resaveS0Block + pyramid via writeDownsampledBlock).
 * SURVEY.md §8(f) row 2. Round-1 scope: input = the dataset's bdv.n5
 * container (the reference's TIFF/CZI ingest readers are out of scope —
 * no image libraries in this environment); output = a new bdv.n5
 * container with per-(setup,timepoint) multi-resolution pyramids
 * "setup{s}/timepoint{t}/s{l}" plus a rewritten dataset.xml pointing at
 * it. The pyramid levels are computed on the GPU (the K8 box-mean
 * kernel via bs_fuse_volume with the single identity view, which is an
 * exact uint16 round-trip at minI=0/maxI=65535). */
#include <atomic>
#include <cstdio>
#include <cstring>
#include <thread>
#include <vector>

#include "../../../include/bigstitch.h"
#include "bs_cli_util.h"
#include "bs_n5.h"
#include "bs_spimdata.h"

int main(int argc, char **argv) {
  bscli::Args args;
  std::map<std::string, std::string> alias = {{"-x", "--xml"},
                                              {"-o", "--n5Path"},
                                              {"-c", "--compression"},
                                              {"-xo", "--xmlOut"}};
  if (!args.parse(argc, argv, alias, {}) || !args.has("xml") ||
      !args.has("n5Path")) {
    fprintf(stderr,
            "usage: resave -x dataset.xml -o out.n5 [-xo out.xml] "
            "[--blockSize 128,128,64] [-c Gzip|Zstandard|Raw] "
            "[--downsamplings \"1,1,1;2,2,1;4,4,2\"] [--device N]\n");
    return 2;
  }
  bssd::SpimData sd;
  std::string err;
  if (!sd.load(args.get("xml"), &err)) {
    fprintf(stderr, "error: %s\n", err.c_str());
    return 1;
  }
  auto bs = bscli::parse_ints(args.get("blockSize", "128,128,64"));
  std::vector<std::array<int, 3>> ladder;
  {
    std::string spec = args.get("downsamplings", "1,1,1;2,2,1");
    std::string cur;
    for (char c : spec + ";") {
      if (c == ';') {
        auto f = bscli::parse_ints(cur);
        if (f.size() == 3)
          ladder.push_back({(int)f[0], (int)f[1], (int)f[2]});
        cur.clear();
      } else {
        cur += c;
      }
    }
  }
  if (ladder.empty() || ladder[0] != std::array<int, 3>{1, 1, 1}) {
    fprintf(stderr, "--downsamplings must start with 1,1,1\n");
    return 2;
  }
  int nlevels = (int)ladder.size();

  bsn5::Container in_n5(sd.n5_path);
  bsn5::Container out_n5(args.get("n5Path"));
  if (!out_n5.create()) {
    fprintf(stderr, "cannot create %s\n", args.get("n5Path").c_str());
    return 1;
  }
  bs_ctx *ctx = nullptr;
  if (bs_ctx_create(&ctx, (int)args.getl("device", 0)) != BS_OK) {
    fprintf(stderr, "error: %s\n", bs_last_error(nullptr));
    return 1;
  }
  std::vector<int32_t> abs_ds(3 * nlevels);
  for (int l = 0; l < nlevels; ++l)
    for (int d = 0; d < 3; ++d) abs_ds[l * 3 + d] = ladder[l][d];

  for (int tp : sd.timepoints) {
    for (auto &s : sd.setups) {
      std::vector<uint16_t> vox;
      std::vector<long long> dims;
      if (!in_n5.read_volume_u16(bssd::SpimData::image_dataset(s.id, tp),
                                 &vox, &dims)) {
        fprintf(stderr, "cannot read view tp=%d setup=%d\n", tp, s.id);
        return 1;
      }
      int64_t d3[3] = {dims[0], dims[1], dims[2]};
      if (bs_view_upload(ctx, s.id, vox.data(), d3) != BS_OK) {
        fprintf(stderr, "upload failed: %s\n", bs_last_error(ctx));
        return 1;
      }
      bs_fuse_view fv{};
      fv.view_id = s.id;
      double ident[12] = {1, 0, 0, 0, 0, 1, 0, 0, 0, 0, 1, 0};
      memcpy(fv.affine, ident, sizeof ident);
      bs_fuse_params prm{};
      prm.fusion_type = BS_FUSION_AVG;
      prm.out_dtype = BS_OUT_UINT16;
      prm.min_intensity = 0;
      prm.max_intensity = 65535;
      prm.interp = 1;
      std::vector<std::vector<char>> hostlvl(nlevels);
      std::vector<void *> lvlptr(nlevels);
      std::vector<int64_t> ldims(3 * nlevels);
      for (int l = 0; l < nlevels; ++l) {
        long long b = 2;
        for (int d = 0; d < 3; ++d)
          b *= (dims[d] + ladder[l][d] - 1) / ladder[l][d];
        hostlvl[l].resize((size_t)b);
        lvlptr[l] = hostlvl[l].data();
      }
      int64_t vmin[3] = {0, 0, 0};
      if (bs_fuse_volume(ctx, &fv, 1, vmin, d3, &prm, nlevels,
                         abs_ds.data(), ldims.data(),
                         lvlptr.data()) != BS_OK) {
        fprintf(stderr, "resave failed: %s\n", bs_last_error(ctx));
        return 1;
      }
      bs_view_release(ctx, s.id);
      for (int l = 0; l < nlevels; ++l) {
        std::string dsn = bssd::SpimData::image_dataset(s.id, tp, l);
        bsn5::DatasetAttrs da;
        da.dims = {ldims[l * 3], ldims[l * 3 + 1], ldims[l * 3 + 2]};
        da.block = {(int)bs[0], (int)bs[1], (int)bs[2]};
        da.dtype = "uint16";
        {
          std::string cn = args.get("compression", "Gzip");
          da.compression = cn == "Zstandard" ? "zstd"
                           : cn == "Raw" ? "raw" : "gzip";
        }
        if (!out_n5.create_dataset(dsn, da)) {
          fprintf(stderr, "cannot create %s\n", dsn.c_str());
          return 1;
        }
        /* downsampling-factors attribute per level (bdv.n5 layout) */
        out_n5.set_attr(dsn, "downsamplingFactors",
                        bsj::Value::mkints(std::vector<int>{
                            ladder[l][0], ladder[l][1], ladder[l][2]}));
        long long lx = ldims[l * 3], ly = ldims[l * 3 + 1],
                  lz = ldims[l * 3 + 2];
        /* independent chunks: compression + file writes on a pool (as
         * the reference's Spark executors write blocks concurrently) */
        const long long ngx = (lx + bs[0] - 1) / bs[0];
        const long long ngy = (ly + bs[1] - 1) / bs[1];
        const long long ngz = (lz + bs[2] - 1) / bs[2];
        const long long nchunks = ngx * ngy * ngz;
        const int NW = (int)std::min<long long>(
            nchunks,
            std::max(1u, std::thread::hardware_concurrency() / 2));
        std::atomic<long long> next(0);
        std::atomic<bool> failed(false);
        auto worker = [&]() {
          std::vector<char> blk((size_t)bs[0] * bs[1] * bs[2] * 2);
          for (;;) {
            long long i = next.fetch_add(1);
            if (i >= nchunks || failed.load()) return;
            const long long gx = i % ngx, gy = (i / ngx) % ngy,
                            gz = i / (ngx * ngy);
            int cx = (int)std::min(bs[0], lx - gx * bs[0]);
            int cy = (int)std::min(bs[1], ly - gy * bs[1]);
            int cz = (int)std::min(bs[2], lz - gz * bs[2]);
            const char *src = hostlvl[l].data();
            for (int z = 0; z < cz; ++z)
              for (int y = 0; y < cy; ++y)
                memcpy(&blk[((size_t)z * cy + y) * cx * 2],
                       src + (((gz * bs[2] + z) * ly + gy * bs[1] + y) *
                                  lx + gx * bs[0]) * 2,
                       (size_t)cx * 2);
            if (!out_n5.write_block(dsn, da, {gx, gy, gz}, blk.data(),
                                    {cx, cy, cz}))
              failed.store(true);
          }
        };
        std::vector<std::thread> ws;
        for (int w = 0; w < NW; ++w) ws.emplace_back(worker);
        for (auto &w : ws) w.join();
        if (failed.load()) {
          fprintf(stderr, "block write failed\n");
          return 1;
        }
      }
      printf("resaved setup %d tp %d: %d level(s)\n", s.id, tp, nlevels);
    }
  }
  bs_ctx_destroy(ctx);
  /* rewritten XML pointing at the new container */
  std::string xo = args.get("xmlOut");
  if (xo.empty()) xo = args.get("xml");
  auto seq = sd.root->child("SequenceDescription");
  auto il = seq ? seq->child("ImageLoader") : nullptr;
  auto n5node = il ? il->child("n5") : nullptr;
  if (n5node) {
    n5node->attrs["type"] = "absolute";
    n5node->text = args.get("n5Path");
  }
  if (!bsx::save_file(xo, sd.root)) {
    fprintf(stderr, "cannot write %s\n", xo.c_str());
    return 1;
  }
  printf("resave done -> %s, %s\n", args.get("n5Path").c_str(), xo.c_str());
  return 0;
}
