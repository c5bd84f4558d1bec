// pybind11 bindings for the daft_amd HIP kernel extension.
#include <torch/extension.h>

#include "api.h"

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "daft_amd hand-written HIP/CDNA4 kernels (gfx950)";
  m.def("hash_rows", &hash_rows, "multi-column 64-bit row hash");
  m.def("compact_indices", &compact_indices, "bool mask -> row indices");
  m.def("take_string", &take_string, "gather string rows");
  m.def("merge_strings", &merge_strings, "if_else merge of string columns");
  m.def("groupby", &groupby, "hash groupby -> (group_ids, rep_idx)");
  m.def("grouped_agg", &grouped_agg, "per-group sum/min/max + valid counts");
  m.def("grouped_count", &grouped_count, "per-group row counts (LDS staged)");
  m.def("dense_first_index", &dense_first_index,
        "first-occurrence index per packed dense key");
  m.def("count_distinct_pairs", &count_distinct_pairs,
        "one-pass per-group distinct-value counts");
  m.def("join_build", &join_build, "bucket-chain hash join build");
  m.def("join_probe", &join_probe, "hash join probe -> (lidx, ridx, matched)");
  m.def("radix_argsort", &radix_argsort, "stable LSD radix argsort of u64");
  m.def("string_chunk_key", &string_chunk_key,
        "big-endian 8-byte chunk keys for string sorting");
  m.def("u64_mod", &u64_mod, "unsigned modulo for hash partitioning");
  m.def("simhash", &simhash);
  m.def("grouped_multi_agg", &grouped_multi_agg);
  m.def("grouped_multi_agg_big", &grouped_multi_agg_big);
  m.def("minhash", &minhash, "word-ngram MinHash signatures (wave/row)");
  m.def("hll_update", &hll_update, "HyperLogLog register updates");
  m.def("image_resize", &image_resize, "bilinear uint8 HWC resize");
  m.def("str_find", &str_find, "substring find/prefix/suffix");
  m.def("str_like", &str_like, "ordered multi-substring LIKE");
  m.def("str_case", &str_case, "ASCII upper/lower");
  m.def("str_substr", &str_substr, "byte substring");
  m.def("str_concat", &str_concat, "row-wise string concat");
  m.def("str_char_length", &str_char_length, "UTF-8 character length");
  m.def("string_compare", &string_compare, "lexicographic compare");
  m.def("levenshtein", &levenshtein,
        "ASCII Levenshtein distance, one thread per row pair");
  m.def("bpe_encode", &bpe_encode,
        "byte-level BPE encode (one wavefront per row, LDS sequence)");
  m.def("fused_jit_available", &fused_jit_available);
  m.def("fused_eval_jit", &fused_eval_jit,
        "hipRTC-compiled straight-line fused expression kernel");
  m.def("fused_eval", &fused_eval,
        "fused stack-interpreter expression evaluation (one kernel per "
        "projection list)");
}
