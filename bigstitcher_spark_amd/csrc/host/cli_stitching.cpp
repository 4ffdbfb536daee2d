/* `stitching` — drop-in for the reference's SparkPairwiseStitching CLI
 * (flag surface: SparkPairwiseStitching.java:76-107; pipeline: :110-393).
 * Host drives libbigstitch's bs_stitch_batch (the computeStitching
 * replacement, :247-255) over all overlapping view pairs, applies the
 * FilteredStitchingResults filters (:369-380: minR/maxR/maxShift*), and
 * writes <StitchingResults> back into dataset.xml (:389).
 *
 * Round-2 surface: view grouping over {illumination, channel} per tile
 * with AVERAGE / PICK_BRIGHTEST aggregation on the GPU ([PIN-GROUP]),
 * the computeStitchingNonEqualTransformations resample fallback
 * ([PIN-NONEQ], :259-267), multi-resolution input levels
 * ([PIN-MIPSEL]), view-selection flags, --fftPadSize POW2|FAST
 * ([PIN-PAD]), and memory-bounded pair windows with view eviction for
 * tile sets larger than HBM. */
#include <cinttypes>
#include <cstdio>
#include <deque>
#include <future>
#include <set>

#include "../../../include/bigstitch.h"
#include "bs_cli_util.h"
#include "bs_imgio.h"
#include "bs_n5.h"
#include "bs_mip.h"
#include "bs_spimdata.h"

using bscli::M34;

int main(int argc, char **argv) {
  bscli::Args args;
  std::map<std::string, std::string> alias = {
      {"-x", "--xml"}, {"-ds", "--downsampling"}, {"-p", "--peaksToCheck"},
      {"-vi", "--vi"}};
  if (!args.parse(argc, argv, alias,
                  {"disableSubpixelResolution", "dryRun",
                   "localSparkBindAddress"}) ||
      !args.has("xml")) {
    fprintf(stderr,
            "usage: stitching -x dataset.xml [-ds 2,2,1] [-p 5] "
            "[--minR 0.3] [--maxR 1.0] [--maxShiftX px] [--maxShiftY px] "
            "[--maxShiftZ px] [--maxShiftTotal px] "
            "[--disableSubpixelResolution] [--channelCombine AVERAGE] "
            "[--illumCombine PICK_BRIGHTEST] "
            "[-vi 'tp,setup' ... | --angleId/--tileId/--channelId/"
            "--illuminationId/--timepointId '0,1,..'] [--device N] "
            "[--dryRun]\n");
    return 2;
  }
  /* Spark-infrastructure flags accepted for drop-in compatibility;
   * meaningless here (no Spark, no S3) */
  for (const char *f : {"s3Region", "localSparkBindAddress"})
    if (args.has(f))
      fprintf(stderr, "note: --%s accepted for compatibility (no-op in "
                      "this build)\n", f);
  const std::string chComb = args.get("channelCombine", "AVERAGE");
  const std::string ilComb = args.get("illumCombine", "PICK_BRIGHTEST");
  for (auto &a : {chComb, ilComb})
    if (a != "AVERAGE" && a != "PICK_BRIGHTEST") {
      fprintf(stderr,
              "unsupported combine action %s (AVERAGE|PICK_BRIGHTEST)\n",
              a.c_str());
      return 2;
    }
  const std::string padSz = args.get("fftPadSize", "POW2");
  if (padSz != "POW2" && padSz != "FAST") {
    fprintf(stderr, "unsupported --fftPadSize %s (POW2|FAST)\n",
            padSz.c_str());
    return 2;
  }
  auto ds = bscli::parse_ints(args.get("downsampling", "2,2,1"));
  if (ds.size() != 3) {
    fprintf(stderr, "bad -ds\n");
    return 2;
  }
  double minR = args.getd("minR", 0.3), maxR = args.getd("maxR", 1.0);
  double maxShift[3] = {args.getd("maxShiftX", 1e300),
                        args.getd("maxShiftY", 1e300),
                        args.getd("maxShiftZ", 1e300)};
  double maxShiftTotal = args.getd("maxShiftTotal", 1e300);

  bssd::SpimData sd;
  std::string err;
  if (!sd.load(args.get("xml"), &err)) {
    fprintf(stderr, "error: %s\n", err.c_str());
    return 1;
  }
  printf("stitching: %zu setups, %zu timepoints, container %s\n",
         sd.setups.size(), sd.timepoints.size(), sd.n5_path.c_str());
  std::vector<bssd::ViewId> selected;
  if (!bssd::select_views(sd, args.getall("vi"), args.get("angleId"),
                          args.get("tileId"), args.get("illuminationId"),
                          args.get("channelId"), args.get("timepointId"),
                          &selected, &err)) {
    fprintf(stderr, "error: %s\n", err.c_str());
    return 1;
  }
  std::set<std::pair<int, int>> selset(selected.begin(), selected.end());

  bsimg::Input n5(sd);
  /* ctx created lazily: --dryRun only enumerates pairs (no GPU) */
  bs_ctx *ctx = nullptr;
  auto ensure_ctx = [&]() -> bool {
    if (ctx) return true;
    if (bs_ctx_create(&ctx, (int)args.getl("device", 0)) != BS_OK) {
      fprintf(stderr, "error: %s\n", bs_last_error(nullptr));
      return false;
    }
    return true;
  };

  std::vector<bssd::StitchEntry> entries;
  for (int tp : sd.timepoints) {
    /* multi-resolution input: per setup pick the largest pyramid level
     * whose factors divide -ds ([PIN-MIPSEL], bs_mip.h) and stitch the
     * remainder; pairs whose two views land on different factors fall
     * back to the largest COMMON dividing level. */
    std::map<int, std::vector<bscli::MipLevel>> setup_levels;
    auto levels_of = [&](int setup) -> const std::vector<bscli::MipLevel> & {
      auto it = setup_levels.find(setup);
      if (it == setup_levels.end())
        it = setup_levels
                 .emplace(setup, n5.read_levels(setup, tp))
                 .first;
      return it->second;
    };
    long long dsv[3] = {ds[0], ds[1], ds[2]};
    auto common_level = [&](int sa, int sb, long long f[3], int *lva,
                            int *lvb) {
      const auto &la = levels_of(sa), &lb = levels_of(sb);
      *lva = *lvb = 0;
      f[0] = f[1] = f[2] = 1;
      long long bestProd = 0;
      for (size_t i = 0; i < la.size(); ++i) {
        bool div = true;
        for (int d = 0; d < 3; ++d)
          if (la[i].f[d] <= 0 || dsv[d] % la[i].f[d] != 0) div = false;
        if (!div) continue;
        int bidx = -1;
        for (size_t j = 0; j < lb.size(); ++j)
          if (lb[j].f[0] == la[i].f[0] && lb[j].f[1] == la[i].f[1] &&
              lb[j].f[2] == la[i].f[2])
            bidx = (int)j;
        long long prod = la[i].f[0] * la[i].f[1] * la[i].f[2];
        if (bidx >= 0 && prod > bestProd) {
          bestProd = prod;
          *lva = (int)i;
          *lvb = bidx;
          for (int d = 0; d < 3; ++d) f[d] = la[i].f[d];
        }
      }
    };
    /* upload views of this timepoint on demand (at a chosen level) */
    std::map<int, int> uploaded; /* setup -> uploaded level */
    std::map<int, std::vector<long long>> updims;
    auto ensure_view = [&](int setup, int level) -> bool {
      auto it = uploaded.find(setup);
      if (it != uploaded.end()) {
        if (it->second != level) {
          fprintf(stderr,
                  "internal: setup %d needed at two pyramid levels\n",
                  setup);
          return false;
        }
        return true;
      }
      std::vector<uint16_t> vox;
      std::vector<long long> dims;
      if (!n5.read_volume_u16(setup, tp, level, &vox, &dims)) {
        fprintf(stderr, "cannot read view tp=%d setup=%d s%d from %s\n",
                tp, setup, level, sd.n5_path.c_str());
        return false;
      }
      int64_t d[3] = {dims[0], dims[1], dims[2]};
      if (bs_view_upload(ctx, setup, vox.data(), d) != BS_OK) {
        fprintf(stderr, "upload failed: %s\n", bs_last_error(ctx));
        return false;
      }
      uploaded[setup] = level;
      updims[setup] = dims;
      return true;
    };

    /* views of one Tile grouped over {Illumination, Channel}; Tiles
     * compared; application over {TimePoint, Angle} — the reference's
     * default SpimDataFilteringAndGrouping (ref :146-161). Setups
     * without attributes form singleton groups. */
    struct VGroup {
      std::vector<const bssd::ViewSetup *> members; /* id-ascending */
      int rep = -1; /* first member: registration + dims source */
    };
    std::vector<VGroup> vgroups;
    {
      std::map<std::pair<int, int>, VGroup> gm;
      int solo = 0;
      for (auto &sv : sd.setups) {
        if (!selset.count({tp, sv.id}) || !sd.regs.count({tp, sv.id}))
          continue;
        auto key = sv.tile >= 0
                       ? std::make_pair(sv.tile, sv.angle)
                       : std::make_pair((1 << 20) + (solo++), -1);
        gm[key].members.push_back(&sv);
      }
      for (auto &kv : gm) {
        std::sort(kv.second.members.begin(), kv.second.members.end(),
                  [](const bssd::ViewSetup *a, const bssd::ViewSetup *b) {
                    return a->id < b->id;
                  });
        kv.second.rep = kv.second.members[0]->id;
        /* grouped members must share the registration (reference TODO
         * at :212 — unequal in-group transforms are not handled there
         * either) */
        for (auto *m : kv.second.members)
          if (!(sd.regs.at({tp, m->id}) ==
                sd.regs.at({tp, kv.second.rep})))
            fprintf(stderr,
                    "warning: grouped views %d and %d have different "
                    "registrations; using setup %d's\n",
                    m->id, kv.second.rep, kv.second.rep);
        vgroups.push_back(kv.second);
      }
    }
    /* enumerate overlapping group pairs (transformed-bbox intersection,
     * the filterNonOverlappingPairs step, ref :165) */
    struct PairPlan {
      int sa, sb; /* representative setups */
      const VGroup *ga = nullptr, *gb = nullptr;
      bs_pair_desc pd;
      M34 ma, mb;
      long long f[3] = {1, 1, 1}; /* shared pyramid-level factors */
      int lva = 0, lvb = 0;       /* level index per view */
      int rem[3] = {1, 1, 1};     /* ds remainder stitched on the GPU */
      bool nonequal = false;      /* resample-to-common-frame fallback */
      double lo[3], hi[3];        /* world overlap box */
      long long rdims[3] = {1, 1, 1}; /* fallback resample grid dims */
    };
    std::vector<PairPlan> plans;
    for (size_t i = 0; i < vgroups.size(); ++i) {
      for (size_t j = i + 1; j < vgroups.size(); ++j) {
        const auto &A = *vgroups[i].members[0], &B = *vgroups[j].members[0];
        auto ra = sd.regs.find({tp, A.id}), rb = sd.regs.find({tp, B.id});
        if (ra == sd.regs.end() || rb == sd.regs.end()) continue;
        double loA[3], hiA[3], loB[3], hiB[3];
        bscli::tbbox(ra->second, A.dims, loA, hiA);
        bscli::tbbox(rb->second, B.dims, loB, hiB);
        bool ovl = true;
        double lo[3], hi[3];
        for (int d = 0; d < 3; ++d) {
          lo[d] = std::max(loA[d], loB[d]);
          hi[d] = std::min(hiA[d], hiB[d]);
          if (hi[d] <= lo[d]) ovl = false;
        }
        if (!ovl) continue;
        if (!bscli::linear_equal(ra->second, rb->second)) {
          /* computeStitchingNonEqualTransformations (ref :259-267)
           * [PIN-NONEQ]: both groups are resampled onto the world
           * overlap box at -ds spacing (inverse-affine trilinear, the
           * fusion sampler) and phase-correlated there; the shift is
           * then in world units of ds. */
          printf("pair (%d,%d): non-translations differ — using "
                 "virtually fused views for stitching\n",
                 A.id, B.id);
          PairPlan pp;
          pp.sa = A.id;
          pp.sb = B.id;
          pp.ga = &vgroups[i];
          pp.gb = &vgroups[j];
          pp.ma = ra->second;
          pp.mb = rb->second;
          pp.nonequal = true;
          for (int d = 0; d < 3; ++d) {
            pp.lo[d] = lo[d];
            pp.hi[d] = hi[d];
            pp.rdims[d] = std::max(
                1LL, (long long)std::ceil((hi[d] - lo[d]) / (double)ds[d]));
          }
          plans.push_back(pp);
          continue;
        }
        /* world overlap -> local interval per view */
        double L[9], Li[9], t[3];
        bscli::decompose(ra->second, L, t);
        if (!bscli::inv3(L, Li)) continue;
        PairPlan pp;
        pp.sa = A.id;
        pp.sb = B.id;
        pp.ga = &vgroups[i];
        pp.gb = &vgroups[j];
        pp.ma = ra->second;
        pp.mb = rb->second;
        pp.pd.view_a = A.id; /* replaced by the combined id at prep */
        pp.pd.view_b = B.id;
        common_level(A.id, B.id, pp.f, &pp.lva, &pp.lvb);
        for (int d = 0; d < 3; ++d)
          pp.rem[d] = (int)(dsv[d] / pp.f[d]);
        if (pp.f[0] * pp.f[1] * pp.f[2] > 1)
          printf("pair (%d,%d): reading pyramid level s%d "
                 "(factors %lld,%lld,%lld, remainder %d,%d,%d)\n",
                 A.id, B.id, pp.lva, pp.f[0], pp.f[1], pp.f[2], pp.rem[0],
                 pp.rem[1], pp.rem[2]);
        bool valid = true;
        for (int v = 0; v < 2; ++v) {
          const auto &m = v == 0 ? ra->second : rb->second;
          const long long *dims = v == 0 ? A.dims : B.dims;
          double tv[3] = {m[3], m[7], m[11]};
          int64_t *off = v == 0 ? pp.pd.off_a : pp.pd.off_b;
          int64_t *size = v == 0 ? pp.pd.size_a : pp.pd.size_b;
          for (int d = 0; d < 3; ++d) {
            /* local = Li (world - t); Li diag-ish handled generally */
            double l0 = 0, l1 = 0;
            for (int k = 0; k < 3; ++k) {
              l0 += Li[d * 3 + k] * (lo[k] - tv[k]);
              l1 += Li[d * 3 + k] * (hi[k] - tv[k]);
            }
            double a = std::min(l0, l1), b = std::max(l0, l1);
            int64_t o = (int64_t)std::floor(a);
            int64_t e = (int64_t)std::ceil(b) + 1;
            if (o < 0) o = 0;
            if (e > dims[d]) e = dims[d];
            /* interval in the uploaded level's grid */
            const long long fd = pp.f[d];
            int64_t lev_dim = (dims[d] + fd - 1) / fd;
            {
              const auto &ls = levels_of(v == 0 ? A.id : B.id);
              int li = v == 0 ? pp.lva : pp.lvb;
              if (li < (int)ls.size() && (int)ls[li].dims.size() == 3)
                lev_dim = ls[li].dims[d];
            }
            int64_t ol = o / fd;
            int64_t el = (e + fd - 1) / fd;
            if (el > lev_dim) el = lev_dim;
            if (el <= ol) valid = false;
            off[d] = ol;
            size[d] = el - ol;
          }
        }
        if (valid) plans.push_back(pp);
      }
    }
    printf("timepoint %d: %zu overlapping pairs\n", tp, plans.size());
    if (args.has("dryRun") || plans.empty()) continue;
    if (!ensure_ctx()) return 1;

    /* prepare each group's stitched image: upload members at the
     * pair's pyramid level and aggregate [PIN-GROUP] (illum action
     * first, then channel action — ref :204-208). Singleton groups
     * pass through. Combined views get ids >= 1<<20. */
    int next_tmp = 1 << 20;
    std::map<std::string, int> combined; /* "rep:level" -> view id */
    auto pick_brightest = [&](const std::vector<int> &ids) -> int {
      int best = ids[0];
      uint64_t bestSum = 0;
      for (size_t q = 0; q < ids.size(); ++q) {
        uint64_t sum = 0;
        if (bs_view_sum(ctx, ids[q], &sum) != BS_OK) return -1;
        if (q == 0 || sum > bestSum) {
          bestSum = sum;
          best = ids[q];
        }
      }
      return best;
    };
    auto combine_ids = [&](std::vector<int> ids,
                           const std::string &action) -> int {
      if (ids.size() == 1) return ids[0];
      if (action == "PICK_BRIGHTEST") return pick_brightest(ids);
      std::vector<int32_t> in(ids.begin(), ids.end());
      int out_id = next_tmp++;
      if (bs_view_combine_avg(ctx, out_id, in.data(), (int)in.size()) !=
          BS_OK) {
        fprintf(stderr, "combine failed: %s\n", bs_last_error(ctx));
        return -1;
      }
      return out_id;
    };
    auto prepare_group = [&](const VGroup &g, int level,
                             int *out_id) -> bool {
      if (g.members.size() == 1) {
        if (!ensure_view(g.rep, level)) return false;
        *out_id = g.rep;
        return true;
      }
      std::string key = std::to_string(g.rep) + ":" + std::to_string(level);
      auto itc = combined.find(key);
      if (itc != combined.end()) {
        *out_id = itc->second;
        return true;
      }
      for (auto *m : g.members)
        if (!ensure_view(m->id, level)) return false;
      /* by channel, illum-ascending */
      std::map<int, std::vector<int>> by_ch;
      for (auto *m : g.members) by_ch[m->channel].push_back(m->id);
      std::vector<int> ch_ids;
      for (auto &kv : by_ch) {
        int cid = combine_ids(kv.second, ilComb);
        if (cid < 0) return false;
        ch_ids.push_back(cid);
      }
      int result = combine_ids(ch_ids, chComb);
      if (result < 0) return false;
      printf("group of %zu views (rep %d): combined into view %d "
             "(illum %s, channel %s)\n",
             g.members.size(), g.rep, result, ilComb.c_str(),
             chComb.c_str());
      combined[key] = result;
      *out_id = result;
      return true;
    };
    /* memory-bounded windows: pairs are prepared and stitched in
     * chunks sized so the resident views (this window's + still-needed
     * earlier ones) stay under the free-HBM budget; views whose last
     * use has passed are released after each chunk. Large tile grids
     * (views >> HBM) stitch on one GPU. */
    uint64_t freeb = 0, totb = 0;
    (void)bs_device_mem(ctx, &freeb, &totb);
    size_t budget = freeb > (8UL << 30) ? (size_t)freeb - (4UL << 30)
                                        : (size_t)freeb * 3 / 4;
    if (const char *e = getenv("BS_STITCH_BUDGET_MB"))
      budget = (size_t)atoll(e) << 20;
    std::map<int, size_t> sbytes;     /* conservative: full-res bytes */
    std::map<int, size_t> last_use;   /* setup -> last plan index */
    for (size_t k = 0; k < plans.size(); ++k)
      for (const VGroup *g2 : {plans[k].ga, plans[k].gb})
        for (auto *m : g2->members) {
          sbytes[m->id] =
              (size_t)m->dims[0] * m->dims[1] * m->dims[2] * 2;
          last_use[m->id] = k;
        }
    std::vector<bs_shift_result> res(plans.size());
    size_t k0 = 0;
    while (k0 < plans.size()) {
      /* grow the window under the budget */
      std::set<int> counted;
      for (auto &kv : uploaded) counted.insert(kv.first);
      size_t bytes = 0;
      for (int id : counted) bytes += sbytes.count(id) ? sbytes[id] : 0;
      size_t k1 = k0;
      while (k1 < plans.size()) {
        size_t add = 0;
        for (const VGroup *g2 : {plans[k1].ga, plans[k1].gb})
          for (auto *m : g2->members)
            if (!counted.count(m->id)) add += sbytes[m->id];
        if (k1 > k0 && bytes + add > budget) break;
        bytes += add;
        for (const VGroup *g2 : {plans[k1].ga, plans[k1].gb})
          for (auto *m : g2->members) counted.insert(m->id);
        ++k1;
      }
      const int tmp_first = next_tmp;
      { /* pre-read this window's views with a one-ahead pipeline: the
           chunk decode of view i+1 overlaps view i's H2D upload (same
           pattern as the fusion CLI band loop) */
        std::vector<std::pair<int, int>> toload; /* (setup, level) */
        std::set<int> seen;
        for (auto &kv : uploaded) seen.insert(kv.first);
        for (size_t k = k0; k < k1; ++k) {
          auto &pp = plans[k];
          const int la = pp.nonequal ? 0 : pp.lva;
          const int lb = pp.nonequal ? 0 : pp.lvb;
          for (auto *m : pp.ga->members)
            if (seen.insert(m->id).second) toload.push_back({m->id, la});
          for (auto *m : pp.gb->members)
            if (seen.insert(m->id).second) toload.push_back({m->id, lb});
        }
        struct RV {
          std::vector<uint16_t> vox;
          std::vector<long long> dims;
          bool ok;
        };
        auto read_one = [&](size_t i) {
          return std::async(std::launch::async, [i, &toload, &n5, tp] {
            RV r;
            r.ok = n5.read_volume_u16(toload[i].first, tp,
                                      toload[i].second, &r.vox, &r.dims);
            return r;
          });
        };
        std::deque<std::future<RV>> fus;
        const size_t DEPTH = 3;
        for (size_t i = 0; i < toload.size() && i < DEPTH; ++i)
          fus.push_back(read_one(i));
        for (size_t i = 0; i < toload.size(); ++i) {
          RV r = fus.front().get();
          fus.pop_front();
          if (i + DEPTH < toload.size())
            fus.push_back(read_one(i + DEPTH));
          if (!r.ok) {
            fprintf(stderr, "cannot read view tp=%d setup=%d s%d\n", tp,
                    toload[i].first, toload[i].second);
            return 1;
          }
          int64_t d[3] = {r.dims[0], r.dims[1], r.dims[2]};
          if (bs_view_upload(ctx, toload[i].first, r.vox.data(), d) !=
              BS_OK) {
            fprintf(stderr, "upload failed: %s\n", bs_last_error(ctx));
            return 1;
          }
          uploaded[toload[i].first] = toload[i].second;
          updims[toload[i].first] = r.dims;
        }
      }
      for (size_t k = k0; k < k1; ++k) {
        auto &pp = plans[k];
      if (pp.nonequal) {
        /* resample both groups onto the overlap box at ds spacing via
         * the fusion sampler, then stitch the resampled volumes 1:1 */
        int ga_id, gb_id;
        if (!prepare_group(*pp.ga, 0, &ga_id) ||
            !prepare_group(*pp.gb, 0, &gb_id))
          return 1;
        for (int side = 0; side < 2; ++side) {
          const M34 &m = side == 0 ? pp.ma : pp.mb;
          bs_fuse_view fv{};
          fv.view_id = side == 0 ? ga_id : gb_id;
          for (int r2 = 0; r2 < 3; ++r2) {
            for (int c2 = 0; c2 < 3; ++c2)
              fv.affine[r2 * 4 + c2] = m[r2 * 4 + c2] / (double)ds[r2];
            fv.affine[r2 * 4 + 3] =
                (m[r2 * 4 + 3] - pp.lo[r2]) / (double)ds[r2];
          }
          bs_fuse_params fprm{};
          fprm.fusion_type = BS_FUSION_AVG;
          fprm.out_dtype = BS_OUT_UINT16;
          fprm.min_intensity = 0;
          fprm.max_intensity = 65535;
          fprm.interp = 1;
          int64_t vmin[3] = {0, 0, 0};
          int64_t vdim[3] = {pp.rdims[0], pp.rdims[1], pp.rdims[2]};
          int32_t l0[3] = {1, 1, 1};
          int64_t ld[3];
          std::vector<uint16_t> host((size_t)pp.rdims[0] * pp.rdims[1] *
                                     pp.rdims[2]);
          void *bufs[1] = {host.data()};
          if (bs_fuse_volume(ctx, &fv, 1, vmin, vdim, &fprm, 1, l0, ld,
                             bufs) != BS_OK) {
            fprintf(stderr, "resample failed: %s\n", bs_last_error(ctx));
            return 1;
          }
          int rid = next_tmp++;
          if (bs_view_upload(ctx, rid, host.data(), vdim) != BS_OK)
            return 1;
          int64_t *off = side == 0 ? pp.pd.off_a : pp.pd.off_b;
          int64_t *size = side == 0 ? pp.pd.size_a : pp.pd.size_b;
          for (int d = 0; d < 3; ++d) {
            off[d] = 0;
            size[d] = pp.rdims[d];
          }
          (side == 0 ? pp.pd.view_a : pp.pd.view_b) = rid;
        }
        continue;
      }
      int ga_id, gb_id;
      if (!prepare_group(*pp.ga, pp.lva, &ga_id) ||
          !prepare_group(*pp.gb, pp.lvb, &gb_id))
        return 1;
      pp.pd.view_a = ga_id;
      pp.pd.view_b = gb_id;
      }
    /* one bs_stitch_batch per distinct ds-remainder (pairs from
     * different pyramid depths stitch at different residual factors) */
    std::map<std::array<int, 3>, std::vector<size_t>> groups;
    for (size_t k = k0; k < k1; ++k)
      groups[{plans[k].rem[0], plans[k].rem[1], plans[k].rem[2]}]
          .push_back(k);
    for (auto &g : groups) {
      std::vector<bs_pair_desc> pds;
      for (size_t k : g.second) pds.push_back(plans[k].pd);
      bs_stitch_params prm{};
      prm.ds[0] = g.first[0];
      prm.ds[1] = g.first[1];
      prm.ds[2] = g.first[2];
      prm.peaks_to_check = (int)args.getl("peaksToCheck", 5);
      prm.do_subpixel = args.has("disableSubpixelResolution") ? 0 : 1;
      prm.min_overlap_ratio = args.getd("minOverlapRatio", 0.25);
      /* [PIN-PAD]: pow2 (default) or the reference dependency's even
       * 7-smooth "fast" pad sizes */
      prm.pad_mode = padSz == "FAST" ? 1 : 0;
      std::vector<bs_shift_result> gres(pds.size());
      int rc =
          bs_stitch_batch(ctx, pds.data(), pds.size(), &prm, gres.data());
      if (rc != BS_OK) {
        fprintf(stderr, "stitch failed: %s\n", bs_last_error(ctx));
        return 1;
      }
      for (size_t i = 0; i < g.second.size(); ++i) {
        /* back to full-resolution local px: the kernel already scaled
           by the remainder; the pyramid level contributes f more */
        for (int d = 0; d < 3; ++d)
          gres[i].shift[d] *= (double)plans[g.second[i]].f[d];
        res[g.second[i]] = gres[i];
      }
      }
      /* chunk epilogue: drop views whose last use has passed, the
       * chunk's temp views (combined / resampled), and the combine
       * cache (later chunks recombine on demand) */
      for (auto it = uploaded.begin(); it != uploaded.end();) {
        auto lu = last_use.find(it->first);
        if (lu == last_use.end() || lu->second < k1) {
          (void)bs_view_release(ctx, it->first);
          it = uploaded.erase(it);
        } else {
          ++it;
        }
      }
      for (int id = tmp_first; id < next_tmp; ++id)
        (void)bs_view_release(ctx, id);
      combined.clear();
      k0 = k1;
    }
    for (size_t k = 0; k < plans.size(); ++k) {
      const auto &pp = plans[k];
      const auto &r = res[k];
      if (!r.valid) {
        printf("pair (%d,%d): no shift found\n", pp.sa, pp.sb);
        continue;
      }
      /* world shift = L x local shift (equal-transform path) or
       * ds x grid shift (the fallback's grid axes ARE world axes);
       * filters per FilteredStitchingResults (ref :369-380) */
      double ws[3];
      if (pp.nonequal) {
        for (int d = 0; d < 3; ++d) ws[d] = r.shift[d] * (double)ds[d];
      } else {
        double L[9], t[3];
        bscli::decompose(pp.ma, L, t);
        for (int d = 0; d < 3; ++d)
          ws[d] = L[d * 3 + 0] * r.shift[0] + L[d * 3 + 1] * r.shift[1] +
                  L[d * 3 + 2] * r.shift[2];
      }
      printf("pair (%d,%d): shift=(%.3f, %.3f, %.3f) r=%.4f\n", pp.sa,
             pp.sb, ws[0], ws[1], ws[2], r.r);
      if (r.r < minR || r.r > maxR) continue;
      if (std::fabs(ws[0]) > maxShift[0] || std::fabs(ws[1]) > maxShift[1] ||
          std::fabs(ws[2]) > maxShift[2])
        continue;
      if (std::sqrt(ws[0] * ws[0] + ws[1] * ws[1] + ws[2] * ws[2]) >
          maxShiftTotal)
        continue;
      bssd::StitchEntry e;
      for (auto *m : pp.ga->members) e.views_a.push_back({tp, m->id});
      for (auto *m : pp.gb->members) e.views_b.push_back({tp, m->id});
      for (int d = 0; d < 3; ++d) e.matrix[d * 4 + 3] = ws[d];
      /* bbox: world overlap of the two transformed views */
      double loA[3], hiA[3], loB[3], hiB[3];
      bscli::tbbox(pp.ma, sd.setup(pp.sa)->dims, loA, hiA);
      bscli::tbbox(pp.mb, sd.setup(pp.sb)->dims, loB, hiB);
      for (int d = 0; d < 3; ++d) {
        e.bbox_min[d] = std::max(loA[d], loB[d]);
        e.bbox_max[d] = std::min(hiA[d], hiB[d]);
      }
      e.r = r.r;
      e.hash = bssd::SpimData::calculate_hash(pp.ma, pp.mb);
      entries.push_back(e);
    }
  }
  bs_ctx_destroy(ctx);
  if (!args.has("dryRun")) {
    sd.set_stitching_results(entries);
    if (!sd.save(sd.xml_path)) {
      fprintf(stderr, "cannot write %s\n", sd.xml_path.c_str());
      return 1;
    }
    printf("wrote %zu stitching results to %s\n", entries.size(),
           sd.xml_path.c_str());
  }
  return 0;
}
