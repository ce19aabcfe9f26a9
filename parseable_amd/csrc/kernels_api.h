// Launch wrappers implemented in kernels.hip.
#pragma once
#include <hip/hip_runtime.h>
#include "dev_types.h"

namespace gpuq {
void launch_lz4_seg(hipStream_t, const uint8_t* raw, uint8_t* dec,
                    const DevSeg*, int n, int32_t* d_err);
void launch_lit_lane(hipStream_t, const uint8_t* raw, uint8_t* dec,
                     const DevLit* lits, int64_t n);
void launch_lit_wave(hipStream_t, const uint8_t* raw, uint8_t* dec,
                     const DevLit* lits, int n);
void launch_lz4_backrefs(hipStream_t, uint8_t* dec, const DevBr*,
                         const DevPageBr*, int n);
void launch_brres_lane(hipStream_t, uint8_t* dec, const DevBrRes*,
                       const DevPiece*, int64_t n);
void launch_brres_inl(hipStream_t, uint8_t* dec, const DevBrInl* recs,
                      int64_t n);
void launch_brres_wave(hipStream_t, uint8_t* dec, const DevBrRes*,
                       const DevPiece*, int n);
void launch_def_levels(hipStream_t, const uint8_t* dec, const DevPage*,
                       const int32_t* ids, int n, uint8_t* valid,
                       uint8_t* null_mask, uint32_t* rowof, uint32_t* rank,
                       uint32_t* present, int32_t* d_err);
void launch_contains_win(hipStream_t, const uint8_t* dec, const DevCWin* wins,
                         int n, const DevPage* pages,
                         const uint16_t* starts_pool, const uint8_t* needle,
                         int nlen, const uint32_t* rowof, uint8_t* mask);
void launch_dict_gid(hipStream_t, const uint8_t* dec, const DevPage*,
                     const int32_t* ids, int n, const int32_t* remap_pool,
                     int32_t* out, uint8_t* valid, const uint32_t* present,
                     int mode, int32_t* d_err);
void launch_dict_i64(hipStream_t, const uint8_t* dec, const DevPage*,
                     const int32_t* ids, int n, const int64_t* dictv_pool,
                     int64_t* out, uint8_t* valid, const uint32_t* present,
                     int mode, int32_t* d_err);
void launch_dict_mask(hipStream_t, const uint8_t* dec, const DevPage*,
                      const int32_t* ids, int n, const uint8_t* lut_pool,
                      uint8_t* mask, int32_t* d_err);
void launch_dict_lut_scr(hipStream_t, const uint8_t* dec, const DevPage*,
                         const int32_t* ids, int n, const uint8_t* lut_pool,
                         uint8_t* scr, const uint32_t* present, int32_t* d_err);
void launch_plain_fixed(hipStream_t, const uint8_t* dec, const DevPage*,
                        const int32_t* ids, int n, int64_t* out, uint8_t* valid,
                        const uint32_t* present, int mode, int32_t* d_err);
void launch_expand(hipStream_t, const uint8_t* dec, const DevPage*,
                   const int32_t* ids, int n, const uint8_t* scr,
                   const uint32_t* rank, const uint8_t* valid, uint8_t* out,
                   int mode);
void launch_delta_i64(hipStream_t, const uint8_t* dec, const DevPage*,
                      const int32_t* ids, int n, int64_t* out, uint8_t* valid,
                      int32_t* d_err);
void launch_cmp_i64(hipStream_t, const int64_t* col, const uint8_t* valid,
                    int64_t lo, int64_t hi, int mode, int hi_excl, int is_f64,
                    uint8_t* mask, int64_t n);
void launch_bin_i64(hipStream_t, const int64_t* col, const uint8_t* valid,
                    int64_t origin, int64_t stride, int64_t min_idx,
                    int32_t nbins, int32_t* out, int64_t n);
void launch_init_table(hipStream_t, uint64_t* table, int32_t n_groups,
                       int n_aggs, const int32_t* d_agg_kind);
void launch_dict_count(hipStream_t, const uint8_t* dec, const DevPage*,
                       const int32_t* ids, int n, const int32_t* remap_pool,
                       uint64_t* table, int32_t n_groups, int n_aggs,
                       int32_t* d_err);
void launch_agg(hipStream_t, const AggArgs&);
void launch_pool_vals(hipStream_t, const uint8_t* dec, const DevPage*,
                      const int32_t* ids, int n, const int64_t* pool,
                      int64_t* out, uint8_t* valid, const uint32_t* present,
                      int mode);
void launch_hash_build(hipStream_t, const uint8_t* dec, const int64_t* refs,
                       const uint8_t* valid, int64_t n_rows, uint64_t* hkeys,
                       int32_t* hgids, int clog2, uint32_t* counter,
                       uint64_t* gid2ref, int32_t gid_cap, int32_t* d_err);
void launch_hash_lookup(hipStream_t, const uint8_t* dec, const int64_t* refs,
                        const uint8_t* valid, int64_t n_rows,
                        const uint64_t* hkeys, const int32_t* hgids, int clog2,
                        int32_t* out_gid);
void launch_pair_build(hipStream_t, const int32_t* a, const int32_t* b,
                       int64_t n_rows, uint64_t* hkeys, int32_t* hgids,
                       int clog2, uint32_t* counter, uint64_t* gid2pair,
                       int32_t gid_cap, int32_t* d_err);
void launch_pair_lookup(hipStream_t, const int32_t* a, const int32_t* b,
                        int64_t n_rows, const uint64_t* hkeys,
                        const int32_t* hgids, int clog2, int32_t* out);
void launch_numhash_build(hipStream_t, const int64_t* vals,
                          const uint8_t* valid, int64_t n_rows,
                          uint64_t* hkeys, int32_t* hgids, int clog2,
                          uint32_t* counter, uint64_t* gid2key,
                          int32_t gid_cap, int is_f64, int32_t* d_err);
void launch_numhash_lookup(hipStream_t, const int64_t* vals,
                           const uint8_t* valid, int64_t n_rows,
                           const uint64_t* hkeys, const int32_t* hgids,
                           int clog2, int is_f64, int32_t* out_gid);
void launch_cmp_str(hipStream_t, const uint8_t* dec, const int64_t* refs,
                    const uint8_t* valid, const uint8_t* lit, uint32_t lit_len,
                    int op, uint8_t* mask, int64_t n_rows);
void launch_ref_lens(hipStream_t, const uint8_t* dec, const uint64_t* refs,
                     int64_t n, uint32_t* lens);
void launch_ref_gather(hipStream_t, const uint8_t* dec, const uint64_t* refs,
                       const uint64_t* offs, int64_t n, uint8_t* out);
void launch_compact(hipStream_t, const uint8_t* mask, const int64_t* key_col,
                    int64_t n_rows, int64_t* out_keys, uint32_t* out_rows,
                    unsigned long long* counter);
void launch_gather_i64(hipStream_t, const uint32_t* rows, int64_t k,
                       const int64_t* col, int64_t* out);
void launch_gather_i32(hipStream_t, const uint32_t* rows, int64_t k,
                       const int32_t* col, int32_t* out);
void launch_gather_u8(hipStream_t, const uint32_t* rows, int64_t k,
                      const uint8_t* col, uint8_t* out);
size_t sort_pairs_desc(hipStream_t, void* d_temp, size_t temp_bytes,
                       const int64_t* keys_in, int64_t* keys_out,
                       const uint32_t* rows_in, uint32_t* rows_out, int64_t n);
}  // namespace gpuq
