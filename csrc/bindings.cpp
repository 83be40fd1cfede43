// Python bindings (torch extension) for the MI355X conv engine.
//
// The thin PyTorch-ROCm front end mandated by BASELINE.json:5 binds here:
// buffers are torch CUDA tensors (their allocator), the iteration loop,
// streams and RCCL communicator live in C++ (csrc/engine.hip).

#include <torch/extension.h>

#include "common.hpp"
#include "engine_api.hpp"

namespace pconv {
// io.cpp
void read_block_file(const std::string&, int64_t, int64_t, int64_t, int64_t,
                     int64_t, char*);
void write_block_file(const std::string&, int64_t, int64_t, int64_t, int64_t,
                      int64_t, int64_t, const char*);
std::pair<int, std::vector<float>> parse_filter_file(const std::string&);
}  // namespace pconv

namespace {

pconv::BlockGeom make_geom(int h, int wPx, int haloPx, int S, int pitchE,
                           int bufh, std::vector<int> nbr) {
  pconv::BlockGeom g;
  g.h = h;
  g.wPx = wPx;
  g.haloPx = haloPx;
  g.S = S;
  g.pitchE = pitchE;
  g.bufh = bufh;
  TORCH_CHECK(nbr.size() == 4, "nbr must be [N,S,W,E]");
  for (int i = 0; i < 4; ++i) g.nbr[i] = nbr[i];
  return g;
}

// Direct single-kernel entry for unit tests: dst_region = conv(src) over
// the given region of a padded (bufh, pitchE) buffer pair.
void conv_region(torch::Tensor src, torch::Tensor dst, torch::Tensor wgt,
                 int64_t channels, int64_t k, int64_t haloPx,
                 std::vector<int64_t> region, bool use_mfma) {
  TORCH_CHECK(src.is_cuda() && dst.is_cuda() && wgt.is_cuda());
  TORCH_CHECK(src.dim() == 2 && dst.dim() == 2);
  TORCH_CHECK(src.is_contiguous() && dst.is_contiguous());
  TORCH_CHECK(wgt.scalar_type() == torch::kFloat32);
  pconv::DType dt;
  switch (src.scalar_type()) {
    case torch::kUInt8: dt = pconv::DType::U8; break;
    case torch::kHalf: dt = pconv::DType::F16; break;
    case torch::kFloat32: dt = pconv::DType::F32; break;
    default: TORCH_CHECK(false, "unsupported dtype");
  }
  pconv::BlockGeom g;
  g.pitchE = (int)src.size(1);
  g.bufh = (int)src.size(0);
  g.haloPx = (int)haloPx;
  g.S = (int)channels;
  pconv::Region r{(int)region[0], (int)region[1], (int)region[2],
                  (int)region[3]};
  // Test-only entry point: launch on the null stream, which serializes
  // with torch's default stream on ROCm (torch work on the tensors has
  // completed before and after this call). No torch stream API — this
  // file builds HIP-native, with no hipify pass (BASELINE.json:5 "no
  // CUDA-compat shims").
  pconv::launch_conv(dt, (int)channels, (int)k, src.data_ptr(), dst.data_ptr(),
                     wgt.data_ptr<float>(), g, r, use_mfma,
                     (hipStream_t) nullptr);
  HIP_CHECK(hipGetLastError());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X-native iterative 2D convolution engine (gfx950)";

  py::class_<pconv::ConvEngine>(m, "ConvEngine")
      .def(py::init<int, int, const std::vector<float>&, int, bool,
                    std::vector<int>, std::vector<int>>(),
           py::arg("dtype"), py::arg("k"), py::arg("filter"),
           py::arg("channels"), py::arg("use_mfma") = false,
           py::arg("dot_weights") = std::vector<int>(),
           py::arg("sep_weights") = std::vector<int>())
      .def("add_block",
           [](pconv::ConvEngine& e, torch::Tensor src, torch::Tensor dst,
              int h, int wPx, int haloPx, int S, std::vector<int> nbr) {
             TORCH_CHECK(src.is_cuda() && dst.is_cuda());
             TORCH_CHECK(src.is_contiguous() && dst.is_contiguous());
             TORCH_CHECK(src.sizes() == dst.sizes());
             auto g = make_geom(h, wPx, haloPx, S, (int)src.size(1),
                                (int)src.size(0), std::move(nbr));
             return e.add_block((uintptr_t)src.data_ptr(),
                                (uintptr_t)dst.data_ptr(), g);
           })
      .def("init_nccl",
           [](pconv::ConvEngine& e, py::bytes uid, int rank, int world) {
             e.init_nccl(std::string(uid), rank, world);
           })
      .def("run", &pconv::ConvEngine::run, py::arg("iters"),
           py::arg("chunk_iters"))
      .def("diff_max", &pconv::ConvEngine::diff_max)
      .def("swapped", &pconv::ConvEngine::swapped)
      .def("set_phase_timing", &pconv::ConvEngine::set_phase_timing)
      .def("kernel_path", &pconv::ConvEngine::kernel_path)
      .def("phase_times", &pconv::ConvEngine::phase_times)
      .def("synchronize", &pconv::ConvEngine::synchronize);

  m.def("get_nccl_unique_id",
        []() { return py::bytes(pconv::ConvEngine::get_unique_id()); });

  m.def("conv_region", &conv_region, "single conv kernel over a region",
        py::arg("src"), py::arg("dst"), py::arg("wgt"), py::arg("channels"),
        py::arg("k"), py::arg("halo"), py::arg("region"),
        py::arg("use_mfma") = false);

  m.def("mfma_probe", [](torch::Tensor A, torch::Tensor B) {
    TORCH_CHECK(A.is_cuda() && B.is_cuda());
    TORCH_CHECK(A.sizes() == torch::IntArrayRef({16, 32}));
    TORCH_CHECK(B.sizes() == torch::IntArrayRef({32, 16}));
    auto D = torch::zeros({16, 16}, A.options());
    pconv::launch_mfma_probe(A.contiguous().data_ptr<float>(),
                             B.contiguous().data_ptr<float>(),
                             D.data_ptr<float>(), (hipStream_t) nullptr);
    HIP_CHECK(hipStreamSynchronize((hipStream_t) nullptr));
    return D;
  });

  m.def("read_block",
        [](const std::string& path, int64_t imageRowBytes, int64_t xByteOff,
           int64_t y0, torch::Tensor out) {
          TORCH_CHECK(!out.is_cuda() && out.is_contiguous() && out.dim() == 2);
          int64_t rowBytes = out.size(1) * out.element_size();
          pconv::read_block_file(path, imageRowBytes, xByteOff, y0, out.size(0),
                                 rowBytes, (char*)out.data_ptr());
        });

  m.def("write_block",
        [](const std::string& path, int64_t imageRowBytes, int64_t totalBytes,
           int64_t xByteOff, int64_t y0, torch::Tensor in) {
          TORCH_CHECK(!in.is_cuda() && in.is_contiguous() && in.dim() == 2);
          int64_t rowBytes = in.size(1) * in.element_size();
          pconv::write_block_file(path, imageRowBytes, totalBytes, xByteOff, y0,
                                  in.size(0), rowBytes, (const char*)in.data_ptr());
        });

  m.def("parse_filter", [](const std::string& text) {
    auto [k, vals] = pconv::parse_filter_file(text);
    return py::make_tuple(k, vals);
  });

  m.attr("HIP_BUILT_FOR") = "gfx950";
}
