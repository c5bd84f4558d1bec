// Host-side launcher declarations for the daft_amd HIP kernels.
#pragma once

#include <torch/extension.h>
#include <vector>

using torch::Tensor;
using OptTensor = c10::optional<Tensor>;

// descriptor packing (kernels.hip)
Tensor pack_descs(const std::vector<int64_t>& tags,
                  const std::vector<Tensor>& datas,
                  const std::vector<OptTensor>& offsets,
                  const std::vector<OptTensor>& validities);

// hashing
Tensor hash_rows(const std::vector<int64_t>& tags,
                 const std::vector<Tensor>& datas,
                 const std::vector<OptTensor>& offsets,
                 const std::vector<OptTensor>& validities, int64_t n,
                 int64_t seed);

// selection
Tensor compact_indices(Tensor mask);
std::vector<Tensor> take_string(Tensor offsets, Tensor bytes, Tensor idx);
Tensor merge_strings(Tensor mask, Tensor t_off, Tensor t_bytes, Tensor f_off,
                     Tensor f_bytes, Tensor out_off);

// groupby
std::vector<Tensor> groupby(Tensor hashes, const std::vector<int64_t>& tags,
                            const std::vector<Tensor>& datas,
                            const std::vector<OptTensor>& offsets,
                            const std::vector<OptTensor>& validities);
std::vector<Tensor> grouped_agg(Tensor group_ids, int64_t num_groups,
                                Tensor values, Tensor valid,
                                const std::string& op);
Tensor grouped_count(Tensor group_ids, int64_t num_groups, Tensor valid);
Tensor dense_first_index(Tensor packed, int64_t rng, int64_t sentinel);
Tensor count_distinct_pairs(Tensor hashes, const std::vector<int64_t>& tags,
                            const std::vector<Tensor>& datas,
                            const std::vector<OptTensor>& offsets,
                            const std::vector<OptTensor>& validities,
                            Tensor gids, int64_t num_groups);

// join
std::vector<Tensor> join_build(Tensor hashes);
std::vector<Tensor> join_probe(
    Tensor table, Tensor next, Tensor probe_hashes, Tensor build_hashes,
    const std::vector<int64_t>& tags_l, const std::vector<Tensor>& data_l,
    const std::vector<OptTensor>& off_l, const std::vector<OptTensor>& val_l,
    const std::vector<int64_t>& tags_r, const std::vector<Tensor>& data_r,
    const std::vector<OptTensor>& off_r, const std::vector<OptTensor>& val_r,
    int64_t mode);

// sort
Tensor radix_argsort(Tensor keys);
Tensor string_chunk_key(Tensor offsets, Tensor bytes, int64_t chunk);

// partition
Tensor u64_mod(Tensor hashes, int64_t n_partitions);

// multimodal (multimodal.hip)
Tensor simhash(Tensor offsets, Tensor bytes, int64_t ngram_size);
std::vector<Tensor> grouped_multi_agg(Tensor gids, int64_t num_groups,
                                      std::vector<Tensor> datas,
                                      std::vector<OptTensor> valids,
                                      std::vector<int64_t> ops);
std::vector<Tensor> grouped_multi_agg_big(Tensor gids, int64_t num_groups,
                                          std::vector<Tensor> datas,
                                          std::vector<OptTensor> valids,
                                          std::vector<int64_t> ops);
Tensor minhash(Tensor offsets, Tensor bytes, int64_t num_hashes,
               int64_t ngram_size, Tensor perm_a, Tensor perm_b);
Tensor hll_update(Tensor hashes, Tensor gids, Tensor valid,
                  int64_t num_groups);
Tensor image_resize(Tensor src_bytes, Tensor src_off, Tensor src_h,
                    Tensor src_w, int64_t channels, int64_t out_h,
                    int64_t out_w);

// fused expression interpreter (fusedexpr.hip)
std::vector<Tensor> fused_eval(Tensor prog, Tensor lits,
                               std::vector<Tensor> cols,
                               std::vector<OptTensor> valids,
                               std::vector<int64_t> codes,
                               std::vector<double> scales,
                               std::vector<int64_t> out_codes,
                               std::vector<double> out_scales,
                               std::vector<int64_t> out_need_valid,
                               int64_t n);

// hipRTC JIT for fused expressions (fusedjit.hip)
bool fused_jit_available();
std::vector<Tensor> fused_eval_jit(const std::string& src,
                                   std::vector<Tensor> cols,
                                   std::vector<OptTensor> valids,
                                   std::vector<int64_t> out_codes,
                                   std::vector<int64_t> out_need_valid,
                                   int64_t n);

// BPE tokenize (bpe.hip)
std::vector<Tensor> bpe_encode(Tensor offsets, Tensor bytes, Tensor byte2id,
                               Tensor table_keys, Tensor table_vals);

// edit distance (editdist.hip)
Tensor levenshtein(Tensor a_off, Tensor a_bytes, Tensor b_off,
                   Tensor b_bytes, int64_t max_len);

// strings (strings.hip)
Tensor str_find(Tensor offsets, Tensor bytes, Tensor pattern, int64_t mode);
Tensor str_like(Tensor offsets, Tensor bytes, Tensor needles, Tensor lens,
                bool anchored_prefix, bool anchored_suffix);
Tensor str_case(Tensor offsets, Tensor bytes, int64_t mode);
std::vector<Tensor> str_substr(Tensor offsets, Tensor bytes, int64_t start,
                               int64_t length);
std::vector<Tensor> str_concat(const std::vector<Tensor>& offsets,
                               const std::vector<Tensor>& bytes);
Tensor str_char_length(Tensor offsets, Tensor bytes);
Tensor string_compare(Tensor a_off, Tensor a_bytes, Tensor b_off,
                      Tensor b_bytes);
