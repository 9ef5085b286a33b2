// COCO RLE mask codec + run-length IoU — native CPU implementation.
//
// Replaces pycocotools' maskApi.c for the evaluation
// path: encode/decode of column-major uncompressed RLE and an IoU that
// walks the RUNS of both masks directly (O(#runs), never materializing
// H*W pixels — the pycocotools rleIou approach).

#include <torch/extension.h>

#include <algorithm>
#include <cstdint>
#include <vector>

namespace {

// mask: [H, W] bool/uint8 tensor -> counts (int64 vector), column-major,
// alternating runs of 0s/1s starting with 0s.
std::vector<int64_t> rle_encode_counts(const torch::Tensor& mask) {
  TORCH_CHECK(mask.dim() == 2, "mask must be [H, W]");
  auto m = mask.to(torch::kBool).t().contiguous();  // column-major walk
  const bool* p = m.data_ptr<bool>();
  const int64_t n = m.numel();
  std::vector<int64_t> counts;
  bool cur = false;
  int64_t run = 0;
  for (int64_t i = 0; i < n; ++i) {
    if (p[i] != cur) {
      counts.push_back(run);
      run = 0;
      cur = p[i];
    }
    ++run;
  }
  counts.push_back(run);
  return counts;
}

torch::Tensor rle_encode(torch::Tensor mask) {
  auto counts = rle_encode_counts(mask);
  return torch::tensor(counts, torch::kLong);
}

torch::Tensor rle_decode(torch::Tensor counts, int64_t h, int64_t w) {
  TORCH_CHECK(counts.dim() == 1, "counts must be 1-D");
  auto c = counts.to(torch::kLong).contiguous();
  const int64_t* p = c.data_ptr<int64_t>();
  auto flat = torch::zeros({h * w}, torch::kBool);
  bool* out = flat.data_ptr<bool>();
  int64_t pos = 0;
  bool val = false;
  for (int64_t i = 0; i < c.numel(); ++i) {
    const int64_t run = p[i];
    if (val) {
      std::fill(out + pos, out + std::min(pos + run, h * w), true);
    }
    pos += run;
    val = !val;
  }
  // Column-major layout -> [H, W].
  return flat.view({w, h}).t().contiguous();
}

// Run-walk intersection of two counts lists (same H*W extent).
int64_t rle_intersection(const int64_t* a, int64_t na, const int64_t* b,
                         int64_t nb) {
  int64_t ia = 0, ib = 0;
  int64_t ra = na ? a[0] : 0, rb = nb ? b[0] : 0;  // remaining run
  bool va = false, vb = false;
  int64_t inter = 0;
  while (ia < na && ib < nb) {
    const int64_t step = std::min(ra, rb);
    if (va && vb) inter += step;
    ra -= step;
    rb -= step;
    if (ra == 0) {
      ++ia;
      if (ia < na) ra = a[ia];
      va = !va;
    }
    if (rb == 0) {
      ++ib;
      if (ib < nb) rb = b[ib];
      vb = !vb;
    }
  }
  return inter;
}

int64_t rle_area(const int64_t* c, int64_t n) {
  int64_t area = 0;
  for (int64_t i = 1; i < n; i += 2) area += c[i];
  return area;
}

// dts/gts: lists of 1-D int64 counts tensors; iscrowd: [G] bool.
torch::Tensor rle_iou_matrix(std::vector<torch::Tensor> dts,
                             std::vector<torch::Tensor> gts,
                             torch::Tensor iscrowd) {
  const int64_t D = dts.size(), G = gts.size();
  auto crowd = iscrowd.to(torch::kBool).contiguous();
  TORCH_CHECK(crowd.numel() == G, "iscrowd must have len(gts)");
  const bool* cr = crowd.data_ptr<bool>();
  auto out = torch::zeros({D, G}, torch::kDouble);
  auto acc = out.accessor<double, 2>();
  std::vector<torch::Tensor> dc(D), gc(G);
  std::vector<int64_t> darea(D), garea(G);
  for (int64_t i = 0; i < D; ++i) {
    dc[i] = dts[i].to(torch::kLong).contiguous();
    darea[i] = rle_area(dc[i].data_ptr<int64_t>(), dc[i].numel());
  }
  for (int64_t j = 0; j < G; ++j) {
    gc[j] = gts[j].to(torch::kLong).contiguous();
    garea[j] = rle_area(gc[j].data_ptr<int64_t>(), gc[j].numel());
  }
  for (int64_t i = 0; i < D; ++i) {
    for (int64_t j = 0; j < G; ++j) {
      if (darea[i] == 0 || garea[j] == 0) continue;
      const int64_t inter = rle_intersection(
          dc[i].data_ptr<int64_t>(), dc[i].numel(),
          gc[j].data_ptr<int64_t>(), gc[j].numel());
      if (inter == 0) continue;
      const double uni = cr[j]
          ? (double)darea[i]
          : (double)(darea[i] + garea[j] - inter);
      acc[i][j] = inter / std::max(uni, 1e-10);
    }
  }
  return out;
}

}  // namespace

void register_rle(pybind11::module_& m) {
  m.def("rle_encode", &rle_encode,
        "mask [H,W] -> column-major RLE counts (int64)");
  m.def("rle_decode", &rle_decode, "counts, h, w -> mask [H,W] bool");
  m.def("rle_iou_matrix", &rle_iou_matrix,
        "run-walk IoU matrix [D,G] over RLE counts lists",
        pybind11::arg("dts"), pybind11::arg("gts"),
        pybind11::arg("iscrowd"));
}
