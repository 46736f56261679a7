// Native payload codec for the Adapter transport hot path.
//
// The reference compresses trajectory payloads with lz4.frame from Python
// (`ctools/utils/file_helper.py:255-302`); this image has no lz4 module and
// Python-level zlib holds the GIL.  This pybind module wraps zlib with the
// GIL RELEASED, so the Adapter's fetch/serve threads and the RL dataloader
// worker keep (de)compressing while the learner's Python loop runs.
#include <pybind11/pybind11.h>
#include <stdexcept>
#include <string>
#include <vector>
#include <zlib.h>

namespace py = pybind11;

static py::bytes codec_compress(py::bytes data, int level) {
  std::string in = data;  // copy under GIL
  uLongf bound = compressBound(in.size());
  std::vector<Bytef> out(bound);
  int rc;
  {
    py::gil_scoped_release release;
    rc = compress2(out.data(), &bound,
                   reinterpret_cast<const Bytef*>(in.data()), in.size(), level);
  }
  if (rc != Z_OK) throw std::runtime_error("zlib compress2 failed");
  return py::bytes(reinterpret_cast<const char*>(out.data()), bound);
}

static py::bytes codec_decompress(py::bytes data, size_t size_hint) {
  std::string in = data;
  size_t cap = size_hint ? size_hint : in.size() * 4 + 1024;
  std::vector<Bytef> out;
  int rc = Z_BUF_ERROR;
  uLongf out_len = 0;
  {
    py::gil_scoped_release release;
    while (rc == Z_BUF_ERROR && cap < (1ull << 34)) {
      out.resize(cap);
      out_len = cap;
      rc = uncompress(out.data(), &out_len,
                      reinterpret_cast<const Bytef*>(in.data()), in.size());
      cap *= 4;
    }
  }
  if (rc != Z_OK) throw std::runtime_error("zlib uncompress failed");
  return py::bytes(reinterpret_cast<const char*>(out.data()), out_len);
}

PYBIND11_MODULE(_native_codec, m) {
  m.def("compress", &codec_compress, py::arg("data"), py::arg("level") = 1,
        "zlib compress with the GIL released");
  m.def("decompress", &codec_decompress, py::arg("data"), py::arg("size_hint") = 0,
        "zlib decompress with the GIL released");
}
