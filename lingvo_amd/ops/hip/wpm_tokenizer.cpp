// Native WPM tokenizer hot path.
//
// MI355X-native equivalent of the reference's tokenizer C++ ops
// (lingvo/core/ops/tokenizer_ops_kernels.cc + core/wpm_encoder.py):
// greedy longest-match wordpiece encoding over UTF-8 bytes, exposed as
// a pybind11 class with a GIL-released, multi-threaded batch API so 8
// GPU ranks can tokenize corpora without serializing on Python.
//
// Semantics match lingvo_amd/core/tokenizers.py WpmTokenizer exactly:
// words are whitespace-split, prefixed with "\xe2\x96\x81" (the '▁'
// sentencepiece word mark), matched greedily longest-first; on a miss
// emit unk and advance ONE UTF-8 codepoint (matching the Python
// per-codepoint scan — matches of valid-UTF-8 pieces always land on
// codepoint boundaries, so byte-level matching is equivalent).

#include <torch/extension.h>

#include "wpm_encoder.h"

using lingvo_amd::WpmEncoder;


void RegisterWpmTokenizer(pybind11::module_& m) {
  pybind11::class_<WpmEncoder>(m, "WpmEncoder")
      .def(pybind11::init<std::vector<std::string>, int64_t>(),
           pybind11::arg("pieces"), pybind11::arg("unk_id"))
      .def("vocab_size", &WpmEncoder::VocabSize)
      .def("encode", &WpmEncoder::Encode, pybind11::arg("text"))
      .def("encode_batch", &WpmEncoder::EncodeBatch, pybind11::arg("lines"),
           pybind11::arg("num_threads") = 4)
      .def("decode", &WpmEncoder::Decode, pybind11::arg("ids"));
}
