/*
 * internal.h — shared host-side structures between cstripe_host.cpp (writer,
 * reader, pruning, combine) and cstripe_gpu.hip (staging + kernels).
 */
#ifndef CSTRIPE_INTERNAL_H
#define CSTRIPE_INTERNAL_H

#include <cstdint>
#include <cstddef>
#include <vector>
#include <string>

#include "../../include/cstripe.h"
#include "format.h"

void cs_set_err(const char *fmt, ...);

struct cs_skipnode {
    csf_skipnode n;
    std::vector<csf_seg> segs;       /* decomp_len MASKED at parse (24-bit) */
    std::vector<uint8_t> seg_modes;  /* CSF_SEGMODE_* per segment */
};

struct cs_stripe_info {
    csf_stripe_meta meta;
    uint32_t file_idx = 0;                       /* which mapped shard file */
    std::vector<uint32_t> group_rows;            /* [chunk] */
    std::vector<std::vector<cs_skipnode>> nodes; /* [col][chunk] */
};

/* one mapped shard file */
struct cs_file {
    int fd = -1;
    const uint8_t *map = nullptr;
    size_t map_size = 0;
};

/* a table = one stripe file OR a directory of shard files (the static
 * shard-group mapping of SURVEY §8e: N shards scanned as one table, one
 * partial out) */
struct cstripe_reader {
    std::vector<cs_file> files;
    csf_footer_head head{};
    std::vector<csf_coldef> cols;
    std::vector<cs_stripe_info> stripes;
    const uint8_t *stripe_base(const cs_stripe_info &st) const
    {
        return files[st.file_idx].map + st.meta.file_offset;
    }
};

struct cs_selchunk {
    uint32_t stripe;
    uint32_t chunk;
};

struct cs_gpu_state;   /* defined in cstripe_gpu.hip */

struct cstripe_scan {
    cstripe_reader *r = nullptr;
    uint64_t cols_mask = 0;
    std::vector<cstripe_pred> preds;
    std::vector<cs_selchunk> sel;     /* surviving chunk groups, scan order */
    int64_t chunk_groups_filtered = 0;
    cs_gpu_state *gpu = nullptr;
    size_t batch_pos = 0;             /* next_batch cursor into sel */
    int last_fused = 0;
    double last_kernel_ms = 0.0;
    double last_decode_ms = 0.0;
    double last_agg_ms = 0.0;
    /* random-access read cache (the reference keeps the current stripe
     * open across ColumnarReadRowByRowNumber calls the same way) */
    int64_t rr_gi = -1;
    std::vector<std::vector<uint8_t>> rr_vals;   /* per col, row-aligned */
    std::vector<std::vector<uint8_t>> rr_nulls;
    uint32_t rr_rows = 0;
    uint64_t rr_first = 0;
};

/* implemented in cstripe_gpu.hip */
/* device chunk-group pruning (SelectedChunkMask on GPU):
 * evaluates the CNF min/max refutation for every chunk in one launch;
 * selected[global chunk index] = 1 to keep. Returns CSTRIPE_OK, or
 * CSTRIPE_ERR_NOGPU when no device is visible (caller falls back to the
 * host loop). preds must be normalized + group-sorted (scan_begin's form). */
int  csgpu_prune(cstripe_reader *r, const std::vector<cstripe_pred> &preds,
                 std::vector<uint8_t> &selected);
int  csgpu_stage(cstripe_scan *s, int device_id);
void csgpu_release(cstripe_scan *s);
uint64_t csgpu_staged_bytes(const cstripe_scan *s);
int  csgpu_agg(cstripe_scan *s, const cstripe_agg_spec *aggs, uint32_t n_aggs,
               const uint32_t *group_cols, uint32_t n_group_cols,
               cstripe_group_result *gr, cstripe_partial *out);
int  csgpu_next_batch(cstripe_scan *s, cstripe_batch *batch);
int  csgpu_fetch_batch(cstripe_scan *s, uint32_t gi, cstripe_batch *batch);

/* device write path: one chunk's columns compressed on the GPU (canonical
 * parses where the data fits, raw copy-back otherwise) */
struct cs_dev_chunk_col {
    std::vector<uint8_t> data;   /* canonical LZ4 stream, or raw bytes */
    uint8_t mode = 0;            /* CSF_SEGMODE_* when canonical, else 0 */
    int64_t min_i = 0, max_i = 0;
    bool canonical = false;
    bool has_min_max = false;
};
int csgpu_compress_chunk(const void *const *dev_vals, const uint8_t *types,
                         uint32_t n_cols, uint32_t rows, uint64_t row_offset,
                         std::vector<cs_dev_chunk_col> &out);

#endif
