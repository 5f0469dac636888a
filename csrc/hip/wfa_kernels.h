// C ABI between the host engine (gpu_ops.cpp, compiled by g++) and the
// gfx950 device code (kernels.hip / sortwin.hip, compiled by hipcc).
#pragma once
#include <cstdint>

// opaque stream handle (hipStream_t)
typedef void* wfa_stream_t;

extern "C" {

// ----- elementwise / generator -----
// device-resident synthetic source: ts=start+i, key=mix(i)%n_keys,
// value by dtype: I64 -> mix%1000, F32 -> uniform[0,1), U16(bf16) -> same
void wfa_gen_batch(wfa_stream_t s, int64_t* ts, uint64_t* key, void* val, int vdt,
                   int64_t n, int64_t start, uint64_t seed, uint64_t n_keys);

// map functor catalog (spec ids; see MapSpec in kernels.hip)
//  1 affine_i64  2 affine_f32  3 affine_bf16  4 square_i64  5 exp_decay_f32
void wfa_map_apply(wfa_stream_t s, int spec, void* col, int dt, int64_t n,
                   double a, double b);

// filter predicate -> flags (1 keep)
//  1 mod_ne_i64(m=a,c=b)  2 gt_f32(thr=a)  3 ge_bf16(thr=a)
void wfa_filter_flags(wfa_stream_t s, int spec, const void* col, int dt, int64_t n,
                      double a, double b, uint32_t* flags);

// stream compaction: scatter rows with flag==1 preserving order.
// cols_in/out: arrays of column base pointers (device-visible), esize bytes each.
// n_cols counts payload cols; ts/key handled as two extra i64/u64 columns.
// d_count receives the kept count (device int64).
void wfa_compact(wfa_stream_t s, int64_t n, const uint32_t* flags,
                 uint32_t* scan_tmp,  // >= (nblocks+1) u32
                 const int64_t* ts_in, int64_t* ts_out,
                 const uint64_t* key_in, uint64_t* key_out,
                 const void* const* cols_in, void* const* cols_out,
                 const int* col_esize, int n_cols, int64_t* d_count);

// ----- key -> dense slot hash table (open addressing, u64 keys) -----
// table_packed: u64[2*table_cap] of interleaved (key, slot) 16 B entries,
// both halves init to ~0; n_slots: device counter of allocated slots.
// slot_out[i] = dense id of key[i].
// value-in-key variant: slot_out[i] = slot << 16 | bf16 value (val is u16)
void wfa_key_to_slot_v(wfa_stream_t s, const uint64_t* key, int64_t n,
                       uint64_t* table_packed, uint32_t* n_slots,
                       int64_t table_cap, uint32_t* slot_out,
                       uint64_t* slot_to_key, const uint16_t* val);
void wfa_key_to_slot(wfa_stream_t s, const uint64_t* key, int64_t n,
                     uint64_t* table_packed, uint32_t* n_slots,
                     int64_t table_cap, uint32_t* slot_out,
                     uint64_t* slot_to_key);
// dense-key fast path: slot = key (user-asserted key < max_keys); val
// non-null packs VIK (slot<<16|bf16)
void wfa_key_dense(wfa_stream_t s, const uint64_t* key, int64_t n,
                   int64_t max_keys, uint32_t* slot_out, uint32_t* n_slots,
                   uint32_t* overflow, const uint16_t* val);
// fused variants: also write the radix sort's pass-0 per-block histogram
// (sort blocking) so wfa_sort_pairs2_ph can skip its first count pass
void wfa_key_to_slot_h(wfa_stream_t s, const uint64_t* key, int64_t n,
                       uint64_t* table_packed, uint32_t* n_slots,
                       int64_t table_cap, uint32_t* slot_out,
                       uint64_t* slot_to_key, const uint16_t* val,
                       uint32_t* hist, int shift);
void wfa_key_dense_h(wfa_stream_t s, const uint64_t* key, int64_t n,
                     int64_t max_keys, uint32_t* slot_out, uint32_t* n_slots,
                     uint32_t* overflow, const uint16_t* val, uint32_t* hist,
                     int shift);
void wfa_iota_u64(wfa_stream_t s, uint64_t* p, int64_t n);
// exchange dest bucketing fused with the dest-sort pass-0 histogram
void wfa_bucket_by_key_h(wfa_stream_t s, const uint64_t* key, int64_t n,
                         int world, uint32_t* dest, uint32_t* hist);

// ----- stable LSD radix sort of (slot, iota idx) pairs, 4-bit digits -----
// bits: how many low bits of slot to sort on. tmp arrays sized n (u32 each).
// hist sized >= 16 * nblocks(n) + 16.  After return: slot_sorted, idx_sorted.
void wfa_sort_pairs(wfa_stream_t s, uint32_t* slot, uint32_t* idx,
                    uint32_t* slot_tmp, uint32_t* idx_tmp, uint32_t* hist,
                    int64_t n, int bits,
                    uint32_t** out_slot, uint32_t** out_idx);
// variant carrying a second 32-bit payload (e.g. f32 value bits) through
// the sort so downstream folds read values coalesced instead of gathering
void wfa_sort_pairs2(wfa_stream_t s, uint32_t* slot, uint32_t* idx,
                     uint32_t* slot_tmp, uint32_t* idx_tmp,
                     uint32_t* val2, uint32_t* val2_tmp, uint32_t* hist,
                     int64_t n, int bits, uint32_t** out_slot,
                     uint32_t** out_idx, uint32_t** out_val2,
                     int implicit_iota, int base_shift);
// wfa_sort_pairs2 with the pass-0 histogram precomputed (wfa_key_dense_h)
void wfa_sort_pairs2_ph(wfa_stream_t s, uint32_t* slot, uint32_t* idx,
                        uint32_t* slot_tmp, uint32_t* idx_tmp,
                        uint32_t* val2, uint32_t* val2_tmp, uint32_t* hist,
                        int64_t n, int bits, uint32_t** out_slot,
                        uint32_t** out_idx, uint32_t** out_val2,
                        int implicit_iota, int base_shift);
int64_t wfa_sort_nblocks(int64_t n);
int64_t wfa_sort_hist_u32(int64_t cap);  // hist scratch size in u32

// gather rows by permutation idx (values f32/i64 + ts)
void wfa_gather(wfa_stream_t s, const uint32_t* idx, int64_t n,
                const void* v_in, void* v_out, int esize,
                const int64_t* ts_in, int64_t* ts_out);

// segment boundaries of sorted slot array: seg_start[j] = first index of
// segment j, seg_slot[j] = its slot; d_nseg = #segments.
void wfa_segments(wfa_stream_t s, const uint32_t* slot_sorted, int64_t n,
                  uint32_t* scan_tmp, uint32_t* seg_start, uint32_t* seg_slot,
                  int64_t* d_nseg, int shr);
// bounded-slot fast path: one read + per-slot boundary table (by_slot,
// max_slots u32 entries) + multi-block ordered compaction (scratch32 >=
// 32 u32, e.g. the sort hist)
void wfa_segments_dense(wfa_stream_t s, const uint32_t* slot_sorted, int64_t n,
                        uint32_t* by_slot, int64_t max_slots,
                        uint32_t* scratch32, uint32_t* seg_start,
                        uint32_t* seg_slot, int64_t* d_nseg, int shr);

// ----- per-batch keyed reduction (Reduce_GPU semantics) -----
// comb: 0 sum 1 min 2 max 3 count ; vdt: dtype of v (F32 or I64 accum f64/i64)
void wfa_segment_reduce(wfa_stream_t s, const uint32_t* seg_start,
                        const uint32_t* seg_slot, const int64_t* d_nseg, int64_t n,
                        const void* v_orig, const uint32_t* idx_sorted,
                        const int64_t* ts_orig, int vdt,
                        int comb, const uint64_t* slot_to_key,
                        uint64_t* out_key, void* out_val, int64_t* out_ts,
                        int64_t* d_out_n);
// wave-per-segment variant (f32 values; one wave cooperatively reduces
// one key's segment — use for batches with large segments)
// ts_last: ts column nondecreasing in row order (Batch::ts_mono) => the
// per-segment ts max is its LAST row; skips the per-row random ts gather
void wfa_segment_reduce_wave(
    wfa_stream_t s, const uint32_t* seg_start, const uint32_t* seg_slot,
    const int64_t* d_nseg, int64_t n, const void* v_orig, int vdt,
    const uint32_t* idx_sorted, const int64_t* ts_orig, int comb, int ts_last,
    const uint64_t* slot_to_key, uint64_t* out_key, void* out_val,
    int64_t* out_ts, int64_t* d_out_n);

// closed-form per-segment window-fire counts -> exclusive output offsets
// (nf) + total (*d_out_n); run BEFORE the fold (reads pristine state)
// st_last non-null: also max-folds each key's last-arrival ts (EOS flush
// emit ts) — fused here to save a launch per batch
void wfa_ffat_fire_offsets(wfa_stream_t s, const uint32_t* seg_start,
                           const uint32_t* seg_slot, const int64_t* d_nseg,
                           int64_t n, int64_t pane_len, int64_t P, int64_t S,
                           const uint32_t* st_fill, const uint32_t* st_head,
                           uint32_t* nf, int64_t* d_out_n,
                           const uint32_t* idx_sorted, const int64_t* ts_orig,
                           int64_t* st_last);

// pane-wave variants: fire offsets also emit the completed-pane scan
// (pane_base + *d_total_panes); the fold then runs (A) one wave per
// completed pane -> temp partials + tail -> st_acc_new, (B) one thread per
// segment replaying the per-PANE ring/wsum/fire machine (1/pane_len of
// the per-tuple serial walk).  Use when pane_len >= 32.
void wfa_ffat_fire_offsets_pw(
    wfa_stream_t s, const uint32_t* seg_start, const uint32_t* seg_slot,
    const int64_t* d_nseg, int64_t n, int64_t pane_len, int64_t P, int64_t S,
    const uint32_t* st_fill, const uint32_t* st_head, uint32_t* nf,
    int64_t* d_out_n, const uint32_t* idx_sorted, const int64_t* ts_orig,
    int64_t* st_last, uint32_t* pane_base, int64_t* d_total_panes);
void wfa_ffat_cb_fold_pw(
    wfa_stream_t s, const uint32_t* seg_start, const uint32_t* seg_slot,
    const int64_t* d_nseg, const int64_t* d_total_panes, int64_t n,
    const void* v_f32, int vdt, const uint32_t* idx_sorted,
    const int64_t* ts_orig, int64_t pane_len, int64_t P, int64_t S, int comb,
    int ring_log2, int64_t* st_count, uint32_t* st_fill, float* st_acc,
    float* st_acc_new, float* ring, uint32_t* st_head, float* st_wsum,
    const uint64_t* slot_to_key, const uint32_t* fire_base,
    const uint32_t* pane_base, float* temp, uint64_t* out_key, float* out_val,
    int64_t* out_ts, int64_t out_cap);

// ----- FFAT/pane sliding-window state machine (CB) -----
// Batched multi-key redesign of the reference's per-key FlatFAT_GPU
// (SURVEY.md §7 step 6): all keys' folds advance in ONE kernel over the
// batch's segments; per-slot state lives in dense arenas.
// State arrays (slot-indexed, cap = max_keys):
//   st_count i64, st_fill u32, st_acc f32, ring f32[cap*ring_sz], st_head u32
// Window: win = P panes of pane_len tuples, fires every S panes.
// comb: 0 sum(invertible running total) 1 min 2 max  (min/max combine ring)
// Output appended to out_* at atomic cursor d_out_n.
void wfa_ffat_cb_fold(wfa_stream_t s, const uint32_t* seg_start,
                      const uint32_t* seg_slot, const int64_t* d_nseg, int64_t n,
                      const void* v_f32, int vdt, const uint32_t* idx_sorted,
                      const int64_t* ts_orig,
                      int64_t pane_len, int64_t P, int64_t S, int comb,
                      int ring_log2,
                      int64_t* st_count, uint32_t* st_fill, float* st_acc,
                      float* ring, uint32_t* st_head, float* st_wsum,
                      const uint64_t* slot_to_key, const uint32_t* fire_base,
                      uint64_t* out_key, float* out_val, int64_t* out_ts,
                      int64_t out_cap);

// fused two-stage CB fold: waves compute pane partials into LDS, wave 0
// runs the window machine per chunk — one launch, no global pane traffic
void wfa_ffat_cb_fold_fused(
    wfa_stream_t s, const uint32_t* seg_start, const uint32_t* seg_slot,
    const int64_t* d_nseg, int64_t n, const void* v_f32, int vdt,
    const uint32_t* idx_sorted, const int64_t* ts_orig, int64_t pane_len,
    int64_t P, int64_t S, int comb, int ring_log2, int64_t* st_count,
    uint32_t* st_fill, float* st_acc, float* ring, uint32_t* st_head,
    float* st_wsum, const uint64_t* slot_to_key, const uint32_t* fire_base,
    uint64_t* out_key, float* out_val, int64_t* out_ts, int64_t out_cap);

// MFMA windowed Gram aggregation: per-key tumbling windows over 16-dim
// f32 vectors; window combine Σ v·vᵀ on the matrix cores
// (v_mfma_f32_16x16x4_f32).  Output: 16 rows per fired window
// (key, gwid, ts, Gram row).  d_out_n (from the fire-offset scan) is
// scaled to rows (×16) by the wrapper.
void wfa_gram_prep(wfa_stream_t s, const float* const* colp,
                   const uint32_t* idx_sorted, int64_t n,
                   uint32_t* inv_scratch, float* staged);
void wfa_gram_fold(wfa_stream_t s, const uint32_t* seg_start,
                   const uint32_t* seg_slot, const int64_t* d_nseg,
                   int64_t n, const float* const* colp,
                   const uint32_t* idx_sorted, const int64_t* ts_orig,
                   int64_t win, uint32_t* st_fill, float* st_acc,
                   uint32_t* st_head, const uint64_t* slot_to_key,
                   const uint32_t* fire_base, uint64_t* out_key,
                   int64_t* out_gwid, float* const* out_colp,
                   int64_t* out_ts, int64_t out_cap, int64_t* d_out_n, uint32_t* inv_scratch,
                   float* staged);

// stateful map/filter: per-key (slot) f64 state, key-order segment walk;
// map writes results to original positions in place, filter fills flags
void wfa_stateful_apply(wfa_stream_t s, const uint32_t* seg_start,
                        const uint32_t* seg_slot, const int64_t* d_nseg,
                        int64_t n, const uint32_t* idx_sorted, void* col,
                        int dt, int spec, int is_filter, double a, double b,
                        double* state, uint32_t* flags);

// ----- time-based keyed windows: pane lift + watermark-driven advance -----
// One round = lift the batch's tuples into per-slot pending pane partials
// (pane id = ts / pane_len, absolute; pend ring of 2^pend_log2 panes), then
// complete every slot's panes up to limit_pane and fire windows.  Call with
// n == 0 (and null segment args) for a pure watermark/EOS advance.
// stage-split TB entries for the hipEvent harness
void wfa_tb_lift_only(wfa_stream_t s, const uint32_t* seg_start,
                      const uint32_t* seg_slot, const int64_t* d_nseg,
                      int64_t n, const void* v_f32, int vdt,
                      const uint32_t* idx_sorted, const int64_t* ts_orig,
                      int64_t pane_len, int64_t P, int64_t S, int comb,
                      int pend_log2, float* pend, int64_t* pend_base,
                      int64_t* last_pane, uint32_t* ignored,
                      uint32_t* overflow, int ts_mono);
void wfa_tb_countscan_only(wfa_stream_t s, const uint32_t* n_slots,
                           int64_t limit_pane, int64_t* pend_base,
                           int64_t* last_pane, uint32_t* st_head, int64_t P,
                           int64_t S, uint32_t* nf, int64_t* d_out_n);
void wfa_tb_advance_only(wfa_stream_t s, const uint32_t* n_slots,
                         int64_t limit_pane, int64_t pane_len, int64_t P,
                         int64_t S, int comb, int ring_log2, int pend_log2,
                         float* pend, int64_t* pend_base, int64_t* last_pane,
                         uint32_t* st_head, float* st_wsum, float* ring,
                         const uint64_t* slot_to_key, uint32_t* nf,
                         uint64_t* out_key, float* out_val, int64_t* out_ts,
                         int64_t out_cap);
void wfa_ffat_tb_round(
    wfa_stream_t s, const uint32_t* seg_start, const uint32_t* seg_slot,
    const int64_t* d_nseg, int64_t n, const void* v_f32, int vdt,
    const uint32_t* idx_sorted, const int64_t* ts_orig, int64_t pane_len,
    int64_t P, int64_t S, int comb, int ring_log2, int pend_log2,
    int64_t limit_pane, float* pend, int64_t* pend_base, int64_t* last_pane,
    uint32_t* st_head, float* st_wsum, float* ring, const uint32_t* n_slots,
    const uint64_t* slot_to_key, uint32_t* nf, uint32_t* ignored,
    uint32_t* overflow, uint64_t* out_key, float* out_val, int64_t* out_ts,
    int64_t out_cap, int64_t* d_out_n, int ts_mono, uint32_t* span_nf,
                       int64_t* pfirst, int64_t* d_total_spans);

// ----- FlatFAT arena path (non-invertible combines over many panes) -----
// Per-slot complete binary tree over ring of 2^ring_log2 pane leaves,
// stored slot-major: tree[slot * 2*R + node].  Incremental leaf update +
// O(log R) range query, one thread per segment, batched over all keys.
void wfa_ffat_tree_fold(wfa_stream_t s, const uint32_t* seg_start,
                        const uint32_t* seg_slot, const int64_t* d_nseg, int64_t n,
                        const void* v_f32, int vdt, const uint32_t* idx_sorted,
                        const int64_t* ts_orig,
                        int64_t pane_len, int64_t P, int64_t S, int comb,
                        int ring_log2,
                        int64_t* st_count, uint32_t* st_fill, float* st_acc,
                        float* tree, uint32_t* st_head,
                        const uint64_t* slot_to_key, const uint32_t* fire_base,
                        uint64_t* out_key, float* out_val, int64_t* out_ts,
                        int64_t out_cap);

// per-slot last-arrival ts (max-folded) — feeds the EOS flush emit ts
void wfa_seg_last_ts(wfa_stream_t s, const uint32_t* seg_start,
                     const uint32_t* seg_slot, const int64_t* d_nseg, int64_t n,
                     const uint32_t* idx_sorted, const int64_t* ts_orig,
                     int64_t* st_last);

// CB/TB EOS partial-window flush (mirrors CPU FfatCpu on_eos): fires every
// remaining open window from the ring/tree-leaf cells + the open pane.
// st_fill/st_acc null => TB (no partial pane).  cells stride/off: ring =
// (R, 0), tree leaves = (2R, R).  nf: u32 scratch >= n_slots+1.  Drive as
// count -> scan (total) -> one fire call per out_cap-sized page.
void wfa_ffat_cb_flush_fire(wfa_stream_t s, const uint32_t* n_slots, int64_t P,
                            int64_t S, int comb, int ring_log2,
                            const uint32_t* st_fill, const float* st_acc,
                            const float* cells, int64_t slot_stride,
                            int64_t cell_off, const uint32_t* st_head,
                            const int64_t* st_last,
                            const uint64_t* slot_to_key, const uint32_t* nf,
                            int64_t out_base, uint64_t* out_key,
                            float* out_val, int64_t* out_ts, int64_t out_cap);

// single-block exclusive scan over per-slot counts + total
void wfa_slot_scan(wfa_stream_t s, uint32_t* nf, const uint32_t* n_slots,
                   int64_t* d_out_n);

// value-independent counters shared with the hiprtc-generated fold path
void wfa_cb_flush_count(wfa_stream_t s, const uint32_t* n_slots, int64_t P,
                        int64_t S, const uint32_t* st_fill,
                        const uint32_t* st_head, uint32_t* nf);
void wfa_tb_count(wfa_stream_t s, const uint32_t* n_slots, int64_t limit_pane,
                  const int64_t* pend_base, const int64_t* last_pane,
                  const uint32_t* st_head, int64_t P, int64_t S, uint32_t* nf);

// ----- misc -----
void wfa_fill_u64(wfa_stream_t s, uint64_t* p, uint64_t v, int64_t n);
void wfa_fill_u32(wfa_stream_t s, uint32_t* p, uint32_t v, int64_t n);
void wfa_fill_f32(wfa_stream_t s, float* p, float v, int64_t n);
void wfa_fill_f32_strided(wfa_stream_t s, float* p, float v, int64_t n,
                          int64_t stride);
// per-branch split flags: flags[i] = (v[i] == b)
void wfa_flags_eq_u32(wfa_stream_t s, const uint32_t* v, int64_t n, uint32_t b,
                      uint32_t* flags);
void wfa_iota_u32(wfa_stream_t s, uint32_t* p, int64_t n);
void wfa_cast(wfa_stream_t s, const void* in, int dt_in, void* out, int dt_out,
              int64_t n);

// keyby bucketing for RCCL all-to-all: dest = mix(key) % world (stable order)
void wfa_bucket_by_key(wfa_stream_t s, const uint64_t* key, int64_t n, int world,
                       uint32_t* dest_out);

// pack [counts..., wm, done] into an i64 metadata vector on device
void wfa_pack_meta(wfa_stream_t s, const uint32_t* counts, int world, int64_t wm,
                   int64_t done, int64_t* meta);

// histogram of small u32 values (per-destination row counts)
void wfa_count_u32(wfa_stream_t s, const uint32_t* v, int64_t n, uint32_t* counts,
                   int n_bins);

// full-row gather by permutation: ts, key and every payload column
void wfa_gather_rows(wfa_stream_t s, const uint32_t* idx, int64_t n,
                     const int64_t* ts_in, int64_t* ts_out, const uint64_t* key_in,
                     uint64_t* key_out, const void* const* cols_in,
                     void* const* cols_out, const int* esize, int nc);

// unkeyed full-batch reduce: one (value, max_ts) per batch, deterministic
// two-stage tree (no atomics); scratch arrays hold >= 512 entries
void wfa_reduce_all(wfa_stream_t s, const void* v, int vdt, const int64_t* ts,
                    int64_t n, int comb, float* scratch, int64_t* scratch_ts,
                    float* out, int64_t* out_ts);

}  // extern "C"
