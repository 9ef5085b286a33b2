// Native unmold: fused bilinear mask resize + threshold + paste.
//
// Replaces the per-detection python loop of MaskRCNN.unmold_detections
// (reference mrcnn/model.py unmold_detections:~2500 + utils.unmold_mask
// :~630): each [mh, mw] float mask is bilinearly resized to its box,
// thresholded at 0.5 and pasted into a zero [H, W] canvas. One parallel
// pass over detections, no intermediate float frames.
//
// Interpolation matches utils/image.py resize (F.interpolate bilinear,
// align_corners=False): src = (dst + 0.5) * in/out - 0.5 clamped at 0,
// neighbor index1 = min(index0 + 1, in - 1).

#include <torch/extension.h>

#include <ATen/Parallel.h>
#include <algorithm>
#include <cstdint>

namespace {

inline void src_coord(int64_t dst, double scale, int64_t in_size,
                      int64_t* i0, int64_t* i1, float* lam) {
  double s = (static_cast<double>(dst) + 0.5) * scale - 0.5;
  if (s < 0) s = 0;
  int64_t lo = static_cast<int64_t>(s);
  if (lo > in_size - 1) lo = in_size - 1;
  *i0 = lo;
  *i1 = lo < in_size - 1 ? lo + 1 : lo;
  *lam = static_cast<float>(s - static_cast<double>(lo));
}

// masks: [N, mh, mw] float32 (detection's own class already selected)
// boxes: [N, 4] int64/int32 pixel coords (y1, x1, y2, x2) in the canvas
// Returns [N, H, W] bool.
torch::Tensor unmold_masks(torch::Tensor masks, torch::Tensor boxes,
                           int64_t H, int64_t W, double threshold) {
  TORCH_CHECK(masks.dim() == 3, "masks must be [N, mh, mw]");
  TORCH_CHECK(boxes.dim() == 2 && boxes.size(1) == 4,
              "boxes must be [N, 4]");
  TORCH_CHECK(masks.size(0) == boxes.size(0), "N mismatch");
  auto m = masks.to(torch::kFloat).contiguous();
  auto b = boxes.to(torch::kLong).contiguous();
  const int64_t N = m.size(0), mh = m.size(1), mw = m.size(2);
  auto out = torch::zeros({N, H, W}, torch::kBool);
  if (N == 0) return out;
  const float* mp = m.data_ptr<float>();
  const int64_t* bp = b.data_ptr<int64_t>();
  bool* op = out.data_ptr<bool>();
  const float thr = static_cast<float>(threshold);

  at::parallel_for(0, N, 1, [&](int64_t begin, int64_t end) {
    for (int64_t i = begin; i < end; ++i) {
      int64_t y1 = bp[i * 4 + 0], x1 = bp[i * 4 + 1];
      int64_t y2 = bp[i * 4 + 2], x2 = bp[i * 4 + 3];
      // Clamp the paste region to the canvas (callers pre-filter
      // degenerate boxes; clamping keeps out-of-range boxes safe).
      const int64_t oh = y2 - y1, ow = x2 - x1;
      if (oh <= 0 || ow <= 0) continue;
      const float* src = mp + i * mh * mw;
      bool* canvas = op + i * H * W;
      const double sy = static_cast<double>(mh) / oh;
      const double sx = static_cast<double>(mw) / ow;
      for (int64_t oy = 0; oy < oh; ++oy) {
        const int64_t cy = y1 + oy;
        if (cy < 0 || cy >= H) continue;
        int64_t r0, r1;
        float ly;
        src_coord(oy, sy, mh, &r0, &r1, &ly);
        const float* row0 = src + r0 * mw;
        const float* row1 = src + r1 * mw;
        bool* orow = canvas + cy * W;
        for (int64_t ox = 0; ox < ow; ++ox) {
          const int64_t cx = x1 + ox;
          if (cx < 0 || cx >= W) continue;
          int64_t c0, c1;
          float lx;
          src_coord(ox, sx, mw, &c0, &c1, &lx);
          const float top = row0[c0] + lx * (row0[c1] - row0[c0]);
          const float bot = row1[c0] + lx * (row1[c1] - row1[c0]);
          const float v = top + ly * (bot - top);
          orow[cx] = v >= thr;
        }
      }
    }
  });
  return out;
}

}  // namespace

void register_unmold(pybind11::module_& m) {
  m.def("unmold_masks", &unmold_masks,
        "fused bilinear resize + threshold + paste of detection masks",
        pybind11::arg("masks"), pybind11::arg("boxes"),
        pybind11::arg("H"), pybind11::arg("W"),
        pybind11::arg("threshold") = 0.5);
}
