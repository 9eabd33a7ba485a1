// _srx_cpu: CPU-side native core (pybind11 + numpy, no torch headers).
// Round 1 contents: MurmurHash3 hashing (StringStore + HashEmbed row hashing,
// SURVEY.md §2.2 N3).  Transition systems (arc-eager parser / BILUO NER,
// SURVEY.md §2.2 N7) live in transitions.cpp, same module.
#include <pybind11/pybind11.h>
#include <pybind11/numpy.h>
#include <pybind11/stl.h>

#include <cstdint>
#include <string>
#include <vector>

#include "murmur3.h"

namespace py = pybind11;

static py::array_t<uint64_t> hash_strings(const std::vector<std::string>& strs) {
  py::array_t<uint64_t> out((py::ssize_t)strs.size());
  auto r = out.mutable_unchecked<1>();
  for (py::ssize_t i = 0; i < (py::ssize_t)strs.size(); i++) {
    r(i) = srx::hash_utf8(strs[i].data(), (int)strs[i].size());
  }
  return out;
}

static uint64_t hash_string(const std::string& s) {
  return srx::hash_utf8(s.data(), (int)s.size());
}

// Thinc Ops.hash contract: (n,) uint64 ids -> (n, 4) uint32 hash lanes.
static py::array_t<uint32_t> hash4(py::array_t<uint64_t, py::array::c_style | py::array::forcecast> ids,
                                   uint32_t seed) {
  py::ssize_t n = ids.shape(0);
  py::array_t<uint32_t> out({n, (py::ssize_t)4});
  auto in = ids.unchecked<1>();
  auto r = out.mutable_unchecked<2>();
  for (py::ssize_t i = 0; i < n; i++) {
    uint32_t h[4];
    srx::murmur3_hash4_u64(in(i), seed, h);
    r(i, 0) = h[0]; r(i, 1) = h[1]; r(i, 2) = h[2]; r(i, 3) = h[3];
  }
  return out;
}

// HashEmbed row ids: (n,) uint64 keys -> (n,4) int32 rows in [0, nrows).
static py::array_t<int32_t> hashembed_rows(py::array_t<uint64_t, py::array::c_style | py::array::forcecast> ids,
                                           uint32_t seed, uint32_t nrows) {
  py::ssize_t n = ids.shape(0);
  py::array_t<int32_t> out({n, (py::ssize_t)4});
  auto in = ids.unchecked<1>();
  auto r = out.mutable_unchecked<2>();
  for (py::ssize_t i = 0; i < n; i++) {
    uint32_t h[4];
    srx::murmur3_hash4_u64(in(i), seed, h);
    r(i, 0) = (int32_t)(h[0] % nrows);
    r(i, 1) = (int32_t)(h[1] % nrows);
    r(i, 2) = (int32_t)(h[2] % nrows);
    r(i, 3) = (int32_t)(h[3] % nrows);
  }
  return out;
}

void init_transitions(py::module_& m);  // transitions.cpp

PYBIND11_MODULE(_srx_cpu, m) {
  m.doc() = "spacy_ray_amd CPU native core (murmur hashing, transition systems)";
  m.def("hash_strings", &hash_strings, "hash a list of UTF-8 strings to uint64");
  m.def("hash_string", &hash_string, "hash one UTF-8 string to uint64");
  m.def("spacy_hash_strings",
        [](const std::vector<std::string>& strs) {
          // spaCy StringStore hash (MurmurHash64A seed 1) — the `.spacy`
          // DocBin string-reference values
          py::array_t<uint64_t> out((py::ssize_t)strs.size());
          auto r = out.mutable_unchecked<1>();
          for (py::ssize_t i = 0; i < (py::ssize_t)strs.size(); i++)
            r(i) = srx::murmur2_64a(strs[i].data(), (int)strs[i].size(), 1);
          return out;
        },
        "spaCy-compatible string hashes (MurmurHash64A, seed 1)");
  m.def("spacy_hash_string", [](const std::string& s) {
    return srx::murmur2_64a(s.data(), (int)s.size(), 1);
  });
  m.def("hash4", &hash4, py::arg("ids"), py::arg("seed"),
        "murmur3 x86_128 of 8-byte keys -> (n,4) uint32");
  m.def("hashembed_rows", &hashembed_rows, py::arg("ids"), py::arg("seed"), py::arg("nrows"),
        "murmur3 row ids for HashEmbed: (n,4) int32 in [0, nrows)");
  init_transitions(m);
}
