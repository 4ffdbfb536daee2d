/* bigstitch.h — C ABI of libbigstitch, the MI355X-native replacement for the
 * math layer of BigStitcher-Spark's two hot paths.
 *
 * Drop-in boundary (see SURVEY.md §8(b) and INTEGRATION.md):
 *
 *   bs_stitch_batch  replaces the in-process call site
 *     net.preibisch.stitcher.algorithm.globalopt.TransformationTools
 *         .computeStitching(groupA, groupB, vrs, params, sd, gva, ds, service)
 *       -> Pair<Pair<AffineGet,Double>, RealInterval>
 *     called from reference SparkPairwiseStitching.java:247-255.
 *     One bs_pair_desc = one tile-pair RDD element (SparkPairwiseStitching.java:192-303).
 *
 *   bs_fuse_blocks  replaces the in-process call site
 *     net.preibisch.mvrecon.process.fusion.blk.BlkAffineFusion
 *         .initWithIntensityCoefficients(conv, imgLoader, views, regs, vds,
 *             fusionType, ..., interp=1, coeffs, bbox, type, blockSize)
 *       -> BlockSupplier<T>   (+ the materializing BlockAlgoUtils.arrayImg copy)
 *     called from reference SparkAffineFusion.java:602-615, :627.
 *     One bs_block_desc = one output-grid element of Grid.create(dims,
 *     computeBlockSize, blockSize) (SparkAffineFusion.java:459-461).
 *
 * Conventions:
 *   - All voxel volumes are uint16, C-contiguous with X fastest:
 *     linear index = x + dims[0]*(y + dims[1]*z); dims = {nx, ny, nz}.
 *     (imglib2 dimension order: dim 0 = x.)
 *   - Affine matrices are row-major 3x4 double, mapping view-local (x,y,z,1)
 *     to world coordinates, identical to the reference's AffineTransform3D
 *     serialization (Spark.java:201-233: double[3][4]).
 *   - Caller owns every buffer. No exceptions cross the ABI; every entry
 *     point returns 0 on success or a negative BS_E* code. Thread-safe per
 *     context. Blocking; batching supplies parallelism.
 *   - A JNI binding for the stock Java host binds these symbols 1:1
 *     (see INTEGRATION.md).
 */
#ifndef BIGSTITCH_H
#define BIGSTITCH_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

#define BS_OK 0
#define BS_EINVAL -1   /* bad argument */
#define BS_ENODEV -2   /* no HIP device / HIP error during init */
#define BS_ENOMEM -3   /* device or host allocation failed */
#define BS_EHIP -4     /* HIP runtime error mid-operation */
#define BS_ENOVIEW -5  /* view_id not uploaded */
#define BS_EUNSUP -6   /* unsupported parameter combination */

typedef struct bs_ctx bs_ctx;

/* Create a context bound to one HIP device (one context per GPU; work units
 * are hash-sharded across contexts by the host, mirroring the reference's
 * independent-RDD-element model — SURVEY.md §5 "no collectives"). */
int bs_ctx_create(bs_ctx **out, int device_id);
void bs_ctx_destroy(bs_ctx *ctx);

/* Last error message for a failed call on this ctx (valid until next call). */
const char *bs_last_error(const bs_ctx *ctx);

/* Free/total device memory of the ctx's GPU (host-side capacity
 * planning: the fusion CLI sizes its z-band view window from this). */
int bs_device_mem(bs_ctx *ctx, uint64_t *free_bytes, uint64_t *total_bytes);

/* ------------------------------------------------------------------ views */

/* Upload a view's voxels to device HBM and register them under view_id.
 * Mirrors the reference's per-task image load (the N5 cell fetch behind
 * GroupedViewAggregator / ImgLoader). 288 GB HBM3E holds whole tile sets
 * resident; upload once, stitch/fuse many. */
int bs_view_upload(bs_ctx *ctx, int32_t view_id, const uint16_t *data,
                   const int64_t dims[3]);
int bs_view_release(bs_ctx *ctx, int32_t view_id);

/* Device-generated synthetic view (bench path only; BASELINE.json §(d)):
 * seeded Gaussian-blob content + noise floor. blob array is packed
 * {cx,cy,cz,sigma,amplitude} float5 per blob, positions in THIS view's
 * local coordinates (may lie outside [0,dims)). */
int bs_view_synth(bs_ctx *ctx, int32_t view_id, const int64_t dims[3],
                  const float *blobs, int32_t n_blobs, uint32_t noise_seed,
                  uint16_t noise_floor, uint16_t noise_amp);

/* Download an uploaded/synth view back to host (tests). */
int bs_view_download(bs_ctx *ctx, int32_t view_id, uint16_t *out);

/* Optional per-view linear intensity coefficients for fusion
 * (reference SparkAffineFusion.java:545-559: Coefficients read per view
 * and applied inside BlkAffineFusion.initWithIntensityCoefficients).
 * ab: 2 * gx*gy*gz floats — the multiplicative plane (a), then the
 * additive plane (b), each a coarse grid covering the view uniformly;
 * the fusion kernel samples the grid trilinearly at cell centers
 * ([PIN-COEFF]) and applies value' = a*value + b before weighting.
 * Passing NULL clears the view's coefficients. */
int bs_view_set_coefficients(bs_ctx *ctx, int32_t view_id, const float *ab,
                             const int32_t grid_dims[3]);

/* View-group aggregation primitives (GroupedViewAggregator restatement
 * [PIN-GROUP]; reference SparkPairwiseStitching.java:204-208 applies
 * illumCombine then channelCombine over the {Illumination, Channel}
 * group of each Tile):
 *  - bs_view_combine_avg: out_id := voxelwise mean of n equal-sized
 *    uploaded views, rounded to nearest (ActionType.AVERAGE);
 *  - bs_view_sum: exact integer voxel sum (the host implements
 *    ActionType.PICK_BRIGHTEST by choosing the member with the highest
 *    mean = sum / voxels; exact u64 sums make the pick deterministic).
 */
int bs_view_combine_avg(bs_ctx *ctx, int32_t out_id, const int32_t *in_ids,
                        int32_t n);
int bs_view_sum(bs_ctx *ctx, int32_t view_id, uint64_t *sum);

/* --------------------------------------------------------------- stitching */

/* One tile-pair work unit. The overlap interval inside each view is computed
 * by the host from the current registrations exactly as the reference does
 * before calling computeStitching (PairwiseStitching operates on the
 * overlapping portions of the two grouped views). off/size are in view-local
 * full-resolution voxels, {x,y,z}. */
typedef struct {
  int32_t view_a, view_b;
  int64_t off_a[3], size_a[3];
  int64_t off_b[3], size_b[3];
} bs_pair_desc;

/* PairwiseStitchingParameters (reference SparkPairwiseStitching.java:200-202:
 * new PairwiseStitchingParameters(0, peaksToCheck(5), doSubpixel, ...);
 * downsampling flags :77 default {2,2,1}). */
typedef struct {
  int32_t ds[3];            /* downsample factors per axis, default {2,2,1} */
  int32_t peaks_to_check;   /* default 5 */
  int32_t do_subpixel;      /* default 1 */
  double min_overlap_ratio; /* min candidate overlap as fraction of the
                               smaller (downsampled) interval; default 0.25 */
  int32_t pad_mode;         /* [PIN-PAD] FFT pad size rule: 0 = next
                               power of two (this build's historical
                               default), 1 = next even 7-smooth "fast"
                               size (the imglib2 FFTMethods family the
                               reference's dependency uses) */
  int32_t _pad;             /* alignment */
} bs_stitch_params;

/* The result contract of computeStitching -> SerializablePairwiseStitchingResult
 * (Spark.java:201-233): a pure translation (3x4 with shift in last column),
 * correlation r, validity. shift[] is in FULL-RESOLUTION pixels of view
 * space, meaning: content of view B's interval matches content of view A's
 * interval displaced by +shift (B(x) ~ A(x - shift) over the overlap). */
typedef struct {
  double shift[3];
  double r;
  int32_t valid; /* 0: no peak passed the overlap test */
} bs_shift_result;

int bs_stitch_batch(bs_ctx *ctx, const bs_pair_desc *pairs, size_t n,
                    const bs_stitch_params *params, bs_shift_result *out);

/* ------------------------------------------------------------------ fusion */

/* FusionType enum — reference SparkAffineFusion.java:124-125 (complete
 * except intensity-coefficient and --masks modes; CLOSEST_PIXEL_WINS
 * picks the view whose inverse-mapped point has the largest min-axis
 * distance to its view border [PIN], ties to the first view). */
#define BS_FUSION_AVG 0
#define BS_FUSION_AVG_BLEND 1
#define BS_FUSION_MAX_INTENSITY 2
#define BS_FUSION_LOWEST_VIEWID_WINS 3  /* first contributing view */
#define BS_FUSION_HIGHEST_VIEWID_WINS 4 /* last contributing view */
#define BS_FUSION_CLOSEST_PIXEL_WINS 5  /* largest border distance */

#define BS_OUT_FLOAT32 0
#define BS_OUT_UINT16 1
#define BS_OUT_UINT8 2

/* One input view participating in fusion. affine maps view-local to world
 * (bbox) coordinates; the library inverts it. Blend parameters mirror
 * mvrecon's Blending (border px, cosine ramp range px) used by
 * FusionType.AVG_BLEND. */
typedef struct {
  int32_t view_id;
  double affine[12]; /* row-major 3x4, view-local -> world */
  float blend_border[3];
  float blend_range[3];
} bs_fuse_view;

/* One output block: world-coordinate min and size ({x,y,z}), as produced by
 * Grid.create (reference SparkAffineFusion.java:459-461 — long[][]{offset,
 * size, gridPos}; gridPos is host-side bookkeeping, not part of the math). */
typedef struct {
  int64_t min[3];
  int64_t size[3];
} bs_block_desc;

typedef struct {
  int32_t fusion_type; /* BS_FUSION_*; reference default AVG_BLEND */
  int32_t out_dtype;   /* BS_OUT_*; conversion per RealUnsignedByte/Short
                          Converter min/max scaling (SparkAffineFusion.java:
                          497-517) */
  double min_intensity, max_intensity;
  int32_t interp; /* 1 = trilinear; only value supported (reference :611) */
  int32_t masks;  /* 1 = write coverage masks instead of fused intensities
                     (--masks; reference SparkAffineFusion.java:112-115,
                     :565-578 + fusion/GenerateComputeBlockMasks.java:
                     85-151): out = type max (uint8 255 / uint16 65535 /
                     float32 1.0) where ANY listed view's inverse affine
                     maps the voxel into [-mask_offset, dim-1+mask_offset]
                     per axis (inclusive), else 0. min/max_intensity
                     scaling does not apply. */
  double mask_offset[3]; /* --maskOffset, raw input px (default 0,0,0) */
} bs_fuse_params;

/* Fuse nb output blocks. views[] lists the views overlapping ANY of the
 * blocks; view_idx_per_block/view_idx_offsets give each block's culled view
 * list (the OverlappingViews.findOverlappingViews result, reference
 * fusion/OverlappingViews.java:28-47), as indices into views[].
 * out_blocks[i] receives prod(blocks[i].size) voxels of out_dtype. */
int bs_fuse_blocks(bs_ctx *ctx, const bs_fuse_view *views, size_t nviews,
                   const bs_block_desc *blocks, size_t nb,
                   const int32_t *view_idx_per_block,
                   const int64_t *view_idx_offsets, /* nb+1 prefix offsets */
                   const bs_fuse_params *params, void **out_blocks);

/* Volume-mode fusion + multi-resolution pyramid (SURVEY.md §8(f) row 1:
 * N5ApiTools.setupMultiResolutionPyramid / writeDownsampledBlock,
 * reference SparkAffineFusion.java:703-782, CreateFusionContainer.java:
 * 351-358). Fuses the whole bounding-box volume on-device (internal
 * block grid + OverlappingViews culling), then computes each pyramid
 * level from the previous by box-mean over the RELATIVE downsampling
 * factors (values rounded to nearest for integer dtypes), and stages
 * every level back into the caller's host buffers through pinned
 * double-buffering.
 *   abs_downsampling: nlevels x 3 ints, level 0 must be {1,1,1}; each
 *     level's factors must be integer multiples of the previous level's.
 *   level_dims_out (optional): nlevels x 3, receives ceil(dim/ds).
 *   level_buffers: caller-owned host buffers, one per level, each
 *     prod(level_dims)*dtype_size bytes. */
int bs_fuse_volume(bs_ctx *ctx, const bs_fuse_view *views, size_t nviews,
                   const int64_t vol_min[3], const int64_t vol_dims[3],
                   const bs_fuse_params *params, int32_t nlevels,
                   const int32_t *abs_downsampling, int64_t *level_dims_out,
                   void **level_buffers);

/* Debug-only: download the PCM volume of the LAST pair processed by
 * bs_stitch_batch on this ctx (out must hold prod(out_dims) floats after
 * a first call with out=NULL is not supported — query dims via the pair
 * geometry). Not part of the drop-in surface; used by parity tooling. */
int bs_debug_pcm(bs_ctx *ctx, float *out, int64_t out_dims[3]);

/* ------------------------------------------------------- instrumentation */

/* Per-kernel timing, HIP-event measured on the launch stream (bench.py
 * roofline evidence: average launch duration of the dominant kernel).
 * Accumulated since the last bs_reset_stats on this ctx. */
enum bs_kernel_id {
  BS_K_DOWNSAMPLE = 0,
  BS_K_FFT_X_FWD,  /* R2C load + x-line FFT */
  BS_K_FFT_Y_FWD,
  BS_K_FFT_Z_FWD,
  BS_K_FFT_Z_INV,  /* fused cross-power normalize + inverse z pass */
  BS_K_FFT_Y_INV,
  BS_K_FFT_X_INV,  /* inverse x pass + C2R (PCM write) */
  BS_K_PEAK,       /* PCM local-maxima top-5 tile scan */
  BS_K_PEAK_MERGE,
  BS_K_CORR,       /* candidate cross-correlation integer sums */
  BS_K_SUBPIX,     /* 7-point PCM gather */
  BS_K_FUSE,       /* inverse-affine trilinear blend fusion */
  BS_K_SYNTH,      /* bench-only synthetic tile render */
  BS_K_PYRAMID,    /* 2x (or general) box-mean pyramid level */
  BS_K_COUNT
};

typedef struct {
  double total_ms[BS_K_COUNT];
  long long launches[BS_K_COUNT];
  double batch_ms;   /* whole-batch event-bracketed time (last batch) */
  long long pairs;   /* pairs processed since reset */
  long long blocks;  /* fusion blocks since reset */
} bs_batch_stats;

int bs_get_stats(bs_ctx *ctx, bs_batch_stats *out);
int bs_reset_stats(bs_ctx *ctx);

#ifdef __cplusplus
}
#endif
#endif /* BIGSTITCH_H */
