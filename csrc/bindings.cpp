// PyTorch bindings for the maskrcnn_amd CDNA4 HIP kernels.
//
// Compiled together with csrc/hip/*.hip by hipcc into the in-tree
// maskrcnn_amd/_C extension (see build_ext.py). Tensors arrive as torch
// "cuda" tensors (ROCm masquerades as CUDA); we pull the raw HIP stream
// from c10::hip.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>
#include <tuple>
#include <unordered_map>

extern "C" {
void roi_align_fwd_launch(const void*, const float*, const int*, void*, int,
                          int, int, int, int, int, long, long, long, long,
                          long, long, long, long, int, bool, bool,
                          hipStream_t);
void roi_align_bwd_launch(const void*, const float*, const int*, float*, int,
                          int, int, int, int, int, long, long, long, long,
                          long, long, long, long, int, bool, bool,
                          hipStream_t);
void roi_align_bwd_det_launch(const void*, const float*, const int*,
                              unsigned long long*, float*, int, int, int,
                              int, int, int, long, long, long, long, long,
                              long, long, long, int, bool, bool, float,
                              hipStream_t);
void fixed_to_float_launch(const unsigned long long*, float*, long, float,
                           hipStream_t);
long nms_workspace_bytes(int N);
void nms_launch(const float*, int, float, int, void*, int*, int*,
                hipStream_t);
void fpn_fuse_launch(const void*, const void*, const float*, const void*,
                     void*, int, int, int, int, int, hipStream_t);
void conv3x3_fwd_launch(const void*, const void*, const float*, void*, int,
                        int, int, int, int, bool, hipStream_t);
void conv3x3_v2_launch(const void*, const void*, const float*, const void*,
                       void*, int, int, int, int, int, bool, hipStream_t);
void conv3x3_v3_launch(const void*, const void*, const float*, const void*,
                       void*, int, int, int, int, int, bool, hipStream_t);
void conv3x3_v4_launch(const void*, const void*, const float*, const void*,
                       void*, int, int, int, int, int, bool, hipStream_t);
void conv3x3_v5_launch(const void*, const void*, const float*, const void*,
                       void*, int, int, int, int, int, bool, hipStream_t);
void conv3x3_wrw_launch(const void*, const void*, float*, int, int, int,
                        int, int, hipStream_t);
void conv3x3_splitk_launch(const void*, const void*, const float*, void*,
                           void*, int, int, int, int, int, int, bool,
                           hipStream_t);
void conv3x3_fp8_launch(const void*, const void*, const float*,
                        const float*, const float*, void*, int, int, int,
                        int, int, bool, hipStream_t);
void detection_prep_launch(const void*, const void*, const void*,
                           const float*, const float*, float*, int*,
                           float*, int, int, int, float, bool,
                           hipStream_t);
void deconv2x2_launch(const void*, const void*, const float*, const void*,
                      void*, int, int, int, int, int, bool, hipStream_t);
void conv3x3_mx_launch(const void*, const void*, const void*, const void*,
                       const float*, const void*, void*, int, int, int,
                       int, int, bool, hipStream_t);
void mx_probe_launch(const void*, const void*, const unsigned int*,
                     const unsigned int*, float*, hipStream_t);
void mx_quant_launch(const void*, void*, void*, long, int, hipStream_t);
}

namespace {

bool is_channels_last(const torch::Tensor& t) {
  return t.is_contiguous(at::MemoryFormat::ChannelsLast);
}

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

torch::Tensor roi_align_forward(torch::Tensor feats, torch::Tensor boxes,
                                torch::Tensor box_ind, int64_t pool_h,
                                int64_t pool_w, int64_t sampling,
                                bool aligned) {
  TORCH_CHECK(feats.is_cuda() && boxes.is_cuda() && box_ind.is_cuda(),
              "roi_align: tensors must be on GPU");
  TORCH_CHECK(feats.dim() == 4, "feats must be [B,C,H,W]");
  TORCH_CHECK(boxes.dim() == 2 && boxes.size(1) == 4, "boxes must be [N,4]");
  TORCH_CHECK(feats.scalar_type() == torch::kFloat ||
                  feats.scalar_type() == torch::kBFloat16,
              "roi_align: float/bf16 only");
  const bool cl = is_channels_last(feats);
  if (!cl) feats = feats.contiguous();
  auto boxes_f = boxes.to(torch::kFloat).contiguous();
  auto ind = box_ind.to(torch::kInt).contiguous();

  const int N = boxes.size(0), C = feats.size(1), H = feats.size(2),
            W = feats.size(3);
  auto out = torch::empty(
      {N, C, pool_h, pool_w},
      feats.options().memory_format(cl ? at::MemoryFormat::ChannelsLast
                                       : at::MemoryFormat::Contiguous));
  const bool bf16 = feats.scalar_type() == torch::kBFloat16;
  const int samp = aligned ? std::max<int>(1, sampling) : 0;
  roi_align_fwd_launch(
      feats.data_ptr(), boxes_f.data_ptr<float>(), ind.data_ptr<int>(),
      out.data_ptr(), N, C, H, W, (int)pool_h, (int)pool_w,
      feats.stride(0), feats.stride(1), feats.stride(2), feats.stride(3),
      out.stride(0), out.stride(1), out.stride(2), out.stride(3), samp, cl,
      bf16, cur_stream());
  return out;
}

torch::Tensor roi_align_backward(torch::Tensor grad_out, torch::Tensor boxes,
                                 torch::Tensor box_ind, int64_t B, int64_t C,
                                 int64_t H, int64_t W, int64_t pool_h,
                                 int64_t pool_w, int64_t sampling,
                                 bool aligned) {
  TORCH_CHECK(grad_out.is_cuda(), "grad_out must be on GPU");
  const bool cl = is_channels_last(grad_out);
  if (!cl) grad_out = grad_out.contiguous();
  auto boxes_f = boxes.to(torch::kFloat).contiguous();
  auto ind = box_ind.to(torch::kInt).contiguous();
  const int N = boxes.size(0);
  auto grad_in = torch::zeros(
      {B, C, H, W},
      grad_out.options().dtype(torch::kFloat)
          .memory_format(cl ? at::MemoryFormat::ChannelsLast
                            : at::MemoryFormat::Contiguous));
  const bool bf16 = grad_out.scalar_type() == torch::kBFloat16;
  const int samp = aligned ? std::max<int>(1, sampling) : 0;
  roi_align_bwd_launch(
      grad_out.data_ptr(), boxes_f.data_ptr<float>(), ind.data_ptr<int>(),
      grad_in.data_ptr<float>(), N, (int)C, (int)H, (int)W, (int)pool_h,
      (int)pool_w, grad_in.stride(0), grad_in.stride(1), grad_in.stride(2),
      grad_in.stride(3), grad_out.stride(0), grad_out.stride(1),
      grad_out.stride(2), grad_out.stride(3), samp, cl, bf16, cur_stream());
  return grad_in;
}

torch::Tensor roi_align_backward_det(torch::Tensor grad_out,
                                     torch::Tensor boxes,
                                     torch::Tensor box_ind, int64_t B,
                                     int64_t C, int64_t H, int64_t W,
                                     int64_t pool_h, int64_t pool_w,
                                     int64_t sampling, bool aligned) {
  // Deterministic-reduction variant: fixed-point i64 atomics (integer
  // addition is associative -> bitwise-stable grads across runs).
  // One host sync (amax) per call — test/debug mode, not the hot path.
  TORCH_CHECK(grad_out.is_cuda(), "grad_out must be on GPU");
  const bool cl = is_channels_last(grad_out);
  if (!cl) grad_out = grad_out.contiguous();
  auto boxes_f = boxes.to(torch::kFloat).contiguous();
  auto ind = box_ind.to(torch::kInt).contiguous();
  const int N = boxes.size(0);
  auto grad_in = torch::zeros(
      {B, C, H, W},
      grad_out.options().dtype(torch::kFloat)
          .memory_format(cl ? at::MemoryFormat::ChannelsLast
                            : at::MemoryFormat::Contiguous));
  if (N == 0) return grad_in;
  auto acc = torch::zeros({B * C * H * W},
                          grad_out.options().dtype(torch::kLong));
  const float amax =
      grad_out.abs().amax().to(torch::kFloat).item<float>();
  if (amax == 0.f) return grad_in;
  // Worst case every contribution (<= amax each) lands on one cell:
  // N*PH*PW*S^2*4 <= ~2^21 for N<=2000 — keep the running sum < 2^62.
  const float scale = exp2f(41.f - ceilf(log2f(amax)));
  const bool bf16 = grad_out.scalar_type() == torch::kBFloat16;
  const int samp = aligned ? std::max<int>(1, sampling) : 0;
  roi_align_bwd_det_launch(
      grad_out.data_ptr(), boxes_f.data_ptr<float>(), ind.data_ptr<int>(),
      (unsigned long long*)acc.data_ptr<int64_t>(),
      grad_in.data_ptr<float>(), N, (int)C, (int)H, (int)W, (int)pool_h,
      (int)pool_w, grad_in.stride(0), grad_in.stride(1), grad_in.stride(2),
      grad_in.stride(3), grad_out.stride(0), grad_out.stride(1),
      grad_out.stride(2), grad_out.stride(3), samp, cl, bf16, scale,
      cur_stream());
  // acc is indexed with grad_in's strides; the flat conversion is
  // layout-agnostic (same linear buffer).
  fixed_to_float_launch((const unsigned long long*)acc.data_ptr<int64_t>(),
                        grad_in.data_ptr<float>(), (long)B * C * H * W,
                        1.f / scale, cur_stream());
  return grad_in;
}

torch::Tensor nms(torch::Tensor boxes, torch::Tensor scores,
                  double iou_thresh) {
  TORCH_CHECK(boxes.is_cuda() && scores.is_cuda(), "nms: GPU tensors only");
  TORCH_CHECK(boxes.dim() == 2 && boxes.size(1) == 4, "boxes [N,4]");
  const int N = boxes.size(0);
  if (N == 0)
    return torch::empty({0}, boxes.options().dtype(torch::kLong));
  auto order = scores.argsort(0, /*descending=*/true);
  auto sorted = boxes.index_select(0, order).contiguous().to(torch::kFloat);

  auto ws = torch::empty({nms_workspace_bytes(N)},
                         boxes.options().dtype(torch::kByte));
  auto keep = torch::empty({N}, boxes.options().dtype(torch::kInt));
  auto count = torch::zeros({1}, boxes.options().dtype(torch::kInt));
  nms_launch(sorted.data_ptr<float>(), N, (float)iou_thresh, N,
             (void*)ws.data_ptr(), keep.data_ptr<int>(),
             count.data_ptr<int>(), cur_stream());
  const int n = count.item<int>();  // one DtoH sync
  return order.index_select(0, keep.slice(0, 0, n).to(torch::kLong));
}

std::tuple<torch::Tensor, torch::Tensor> nms_padded(torch::Tensor boxes,
                                                    torch::Tensor scores,
                                                    double iou_thresh,
                                                    int64_t max_out) {
  // Fixed-shape NMS: keep indices (ORIGINAL order) padded with -1 to
  // max_out + the kept count as a DEVICE tensor — zero host syncs, so
  // callers can stay capture-safe (SURVEY §7 hard-part 3).
  TORCH_CHECK(boxes.is_cuda() && scores.is_cuda(),
              "nms_padded: GPU tensors only");
  const int N = boxes.size(0);
  auto count = torch::zeros({1}, boxes.options().dtype(torch::kInt));
  if (N == 0) {
    auto empty = torch::full({max_out}, -1,
                             boxes.options().dtype(torch::kLong));
    return {empty, count};
  }
  auto order = scores.argsort(0, /*descending=*/true);
  auto sorted = boxes.index_select(0, order).contiguous().to(torch::kFloat);
  auto ws = torch::empty({nms_workspace_bytes(N)},
                         boxes.options().dtype(torch::kByte));
  auto keep = torch::zeros({std::max<int64_t>(max_out, 1)},
                           boxes.options().dtype(torch::kInt));
  nms_launch(sorted.data_ptr<float>(), N, (float)iou_thresh, (int)max_out,
             (void*)ws.data_ptr(), keep.data_ptr<int>(),
             count.data_ptr<int>(), cur_stream());
  auto valid = torch::arange(max_out, boxes.options().dtype(torch::kLong))
                   .lt(count.to(torch::kLong));
  auto orig = order.index_select(
      0, keep.slice(0, 0, max_out).to(torch::kLong));
  auto padded = torch::where(
      valid, orig, torch::full_like(orig, -1));
  return {padded, count};
}

torch::Tensor fpn_fuse(torch::Tensor x, torch::Tensor w, torch::Tensor bias,
                       c10::optional<torch::Tensor> up) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "x must be [B,C,H,W] cuda");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "x must be bf16");
  TORCH_CHECK(is_channels_last(x), "x must be channels_last (NHWC)");
  const int B = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
  auto w2 = w.reshape({w.size(0), -1}).contiguous().to(torch::kBFloat16);
  const int Cout = w2.size(0);
  TORCH_CHECK(w2.size(1) == Cin, "weight Cin mismatch");
  TORCH_CHECK(Cin % 32 == 0, "Cin must be a multiple of 32");
  auto bias_f = bias.to(torch::kFloat).contiguous();
  const void* up_ptr = nullptr;
  torch::Tensor up_t;
  if (up.has_value()) {
    up_t = up.value();
    TORCH_CHECK(is_channels_last(up_t) &&
                    up_t.scalar_type() == torch::kBFloat16,
                "up must be channels_last bf16");
    TORCH_CHECK(up_t.size(1) == Cout && up_t.size(2) == H / 2 &&
                    up_t.size(3) == W / 2,
                "up must be [B,Cout,H/2,W/2]");
    up_ptr = up_t.data_ptr();
  }
  auto out = torch::empty(
      {B, Cout, H, W},
      x.options().memory_format(at::MemoryFormat::ChannelsLast));
  fpn_fuse_launch(x.data_ptr(), w2.data_ptr(), bias_f.data_ptr<float>(),
                  up_ptr, out.data_ptr(), B, H, W, Cin, Cout,
                  cur_stream());
  return out;
}

torch::Tensor conv3x3_fwd(torch::Tensor x, torch::Tensor w_packed,
                          torch::Tensor bias, bool relu,
                          int64_t version = 0) {
  // x: [B,Cin,H,W] channels_last bf16; w_packed: [9, Cout, Cin] bf16
  // (pre-transposed tap-major by the python wrapper); bias: [Cout] f32.
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "x must be [B,C,H,W] cuda");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "x must be bf16");
  TORCH_CHECK(is_channels_last(x), "x must be channels_last (NHWC)");
  TORCH_CHECK(w_packed.dim() == 3 && w_packed.size(0) == 9,
              "w_packed must be [9, Cout, Cin]");
  TORCH_CHECK(w_packed.scalar_type() == torch::kBFloat16 &&
                  w_packed.is_contiguous(),
              "w_packed must be contiguous bf16");
  const int B = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
  const int Cout = w_packed.size(1);
  TORCH_CHECK(w_packed.size(2) == Cin, "weight Cin mismatch");
  TORCH_CHECK(Cin % 32 == 0, "Cin must be a multiple of 32");
  auto bias_f = bias.to(torch::kFloat).contiguous();
  auto out = torch::empty(
      {B, Cout, H, W},
      x.options().memory_format(at::MemoryFormat::ChannelsLast));
  const long M = (long)B * H * W;
  int ver = (int)version;
  if (ver == 0)  // auto (r2_conv_ab2 A/B):
    // v5 (256² tile, 8 waves, st_16x32 swizzle) wins at very large M:
    // 725 TF vs MIOpen's ~684 @(2,256,256²,256), 761 vs 740 @Cout=512
    // (r7_conv_ab). v3 wins mid-M (621 vs 459 @(400,256,14²,256) —
    // the 256-row tiles leave that grid at ~1 WG/CU). v1 below 32768.
    ver = (M >= 131072 && Cout >= 128) ? 5
        : (M >= 32768 && Cout >= 128) ? 3 : 1;
  if (ver == 6) {
    // Split-K for fill-bound shapes (M tiles x N tiles < #CUs): the
    // 144 (tap, k-chunk) steps spread over grid.z into per-z fp32
    // slabs, reduced in fixed order (deterministic, no atomics).
    // workspace = split*M*N f32 (16 MB at the target shape).
    const int split = 8;  // (2,512,32²,512) -> 32x8x8 = 2048 blocks
    auto ws = torch::empty({(long)split * M * Cout},
                           x.options().dtype(torch::kFloat));
    conv3x3_splitk_launch(x.data_ptr(), w_packed.data_ptr(),
                          bias_f.data_ptr<float>(), ws.data_ptr(),
                          out.data_ptr(), B, H, W, Cin, Cout, split,
                          relu, cur_stream());
    return out;
  }
  if (ver >= 2) {
    // Persistent 64-B zero page per device (glds cannot predicate
    // out-of-range lanes; they DMA from here instead).
    static std::unordered_map<int, torch::Tensor> zero_pages;
    const int dev = x.get_device();
    auto it = zero_pages.find(dev);
    if (it == zero_pages.end()) {
      it = zero_pages.emplace(
          dev, torch::zeros({32}, x.options())).first;
    }
    auto launch = (ver == 5) ? conv3x3_v5_launch
                : (ver == 4) ? conv3x3_v4_launch
                : (ver == 3) ? conv3x3_v3_launch
                             : conv3x3_v2_launch;
    launch(x.data_ptr(), w_packed.data_ptr(), bias_f.data_ptr<float>(),
           it->second.data_ptr(), out.data_ptr(), B, H, W, Cin, Cout,
           relu, cur_stream());
  } else {
    conv3x3_fwd_launch(x.data_ptr(), w_packed.data_ptr(),
                       bias_f.data_ptr<float>(), out.data_ptr(), B, H, W,
                       Cin, Cout, relu, cur_stream());
  }
  return out;
}

torch::Tensor deconv2x2(torch::Tensor x, torch::Tensor w_packed,
                        torch::Tensor bias, bool relu) {
  // x: [B,Cin,H,W] channels_last bf16; w_packed: [4, Cout, Cin] bf16
  // tap-major (tap = di*2+dj); bias: [Cout] f32. Out: [B,Cout,2H,2W]
  // channels_last (2x2 stride-2 transposed conv).
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "x must be [B,C,H,W] cuda");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "x must be bf16");
  TORCH_CHECK(is_channels_last(x), "x must be channels_last (NHWC)");
  TORCH_CHECK(w_packed.dim() == 3 && w_packed.size(0) == 4,
              "w_packed must be [4, Cout, Cin]");
  TORCH_CHECK(w_packed.scalar_type() == torch::kBFloat16 &&
                  w_packed.is_contiguous(),
              "w_packed must be contiguous bf16");
  const int B = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
  const int Cout = w_packed.size(1);
  TORCH_CHECK(w_packed.size(2) == Cin, "weight Cin mismatch");
  TORCH_CHECK(Cin % 32 == 0, "Cin must be a multiple of 32");
  auto bias_f = bias.to(torch::kFloat).contiguous();
  auto out = torch::empty(
      {B, Cout, 2 * H, 2 * W},
      x.options().memory_format(at::MemoryFormat::ChannelsLast));
  static std::unordered_map<int, torch::Tensor> zero_pages;
  const int dev = x.get_device();
  auto it = zero_pages.find(dev);
  if (it == zero_pages.end())
    it = zero_pages.emplace(dev, torch::zeros({32}, x.options())).first;
  deconv2x2_launch(x.data_ptr(), w_packed.data_ptr(),
                   bias_f.data_ptr<float>(), it->second.data_ptr(),
                   out.data_ptr(), B, H, W, Cin, Cout, relu,
                   cur_stream());
  return out;
}

std::tuple<torch::Tensor, torch::Tensor> mx_quant(torch::Tensor x) {
  // x: [B,C,H,W] bf16 channels_last (C % 64 == 0). Returns
  // (xq [B,H,W,C] u8 e4m3, xs [B,H,W,C/64] u8 e8m0) — the conv3x3_mx
  // input format — in one fused pass.
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
              x.scalar_type() == torch::kBFloat16, "bf16 [B,C,H,W]");
  TORCH_CHECK(is_channels_last(x), "channels_last required");
  const int B = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(C % 64 == 0, "C must be a multiple of 64");
  const long M = (long)B * H * W;
  const int KS = C / 64;
  auto q = torch::empty({B, H, W, C}, x.options().dtype(torch::kByte));
  auto s = torch::empty({B, H, W, KS},
                        x.options().dtype(torch::kByte));
  if (M)
    mx_quant_launch(x.data_ptr(), q.data_ptr(), s.data_ptr(), M * KS,
                    KS, cur_stream());
  return {q, s};
}

torch::Tensor mx_probe(torch::Tensor a, torch::Tensor b,
                       torch::Tensor as, torch::Tensor bs) {
  // a,b: [16,128] u8 e4m3; as,bs: [64] int32 — the FULL per-lane scale
  // registers. Returns [64,4] f32 (each lane's accumulators).
  // Layout probe — test-only.
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && b.is_contiguous() &&
              as.is_contiguous() && bs.is_contiguous(), "contig cuda");
  TORCH_CHECK(as.numel() == 64 && bs.numel() == 64 &&
              as.scalar_type() == torch::kInt &&
              bs.scalar_type() == torch::kInt, "scales: [64] int32");
  auto c = torch::zeros({64, 4}, a.options().dtype(torch::kFloat));
  mx_probe_launch(a.data_ptr(), b.data_ptr(),
                  (const unsigned int*)as.data_ptr<int>(),
                  (const unsigned int*)bs.data_ptr<int>(),
                  c.data_ptr<float>(), cur_stream());
  return c;
}

torch::Tensor conv3x3_mx(torch::Tensor xq, torch::Tensor xs,
                         torch::Tensor wq, torch::Tensor ws,
                         torch::Tensor bias, bool relu) {
  // MX block-scaled e4m3 conv (csrc/hip/conv3x3_mx.hip).
  // xq: [B,H,W,Cin] uint8 contiguous (NHWC bytes);
  // xs: [B,H,W,Cin/32] uint8 e8m0; wq: [9,Cout,Cin] uint8 tap-major;
  // ws: [9,Cout,Cin/32] uint8. Returns [B,Cout,H,W] bf16 channels_last.
  TORCH_CHECK(xq.is_cuda() && xq.dim() == 4 && xq.is_contiguous(),
              "xq must be contiguous [B,H,W,Cin] u8 cuda");
  TORCH_CHECK(xq.scalar_type() == torch::kByte &&
              xs.scalar_type() == torch::kByte &&
              wq.scalar_type() == torch::kByte &&
              ws.scalar_type() == torch::kByte, "u8 tensors required");
  TORCH_CHECK(xs.is_contiguous() && wq.is_contiguous() &&
              ws.is_contiguous(), "contiguous required");
  const int B = xq.size(0), H = xq.size(1), W = xq.size(2),
            Cin = xq.size(3);
  TORCH_CHECK(Cin % 128 == 0, "Cin must be a multiple of 128 (BK)");
  TORCH_CHECK(wq.dim() == 3 && wq.size(0) == 9 && wq.size(2) == Cin,
              "wq must be [9, Cout, Cin]");
  const int Cout = wq.size(1);
  TORCH_CHECK(xs.size(3) == Cin / 64 && ws.size(2) == Cin / 64,
              "scale tensors must have Cin/64 blocks (the gfx950 "
              "mfma_scale 16x16x128 applies one e8m0 per 64 k at "
              "OPSEL=0 — probe-measured)");
  auto bias_f = bias.to(torch::kFloat).contiguous();
  auto out = torch::empty(
      {B, Cout, H, W},
      xq.options().dtype(torch::kBFloat16)
          .memory_format(at::MemoryFormat::ChannelsLast));
  static std::unordered_map<int, torch::Tensor> zero_pages;
  const int dev = xq.get_device();
  auto it = zero_pages.find(dev);
  if (it == zero_pages.end())
    it = zero_pages.emplace(
        dev, torch::zeros({64}, xq.options())).first;
  conv3x3_mx_launch(xq.data_ptr(), xs.data_ptr(), wq.data_ptr(),
                    ws.data_ptr(), bias_f.data_ptr<float>(),
                    it->second.data_ptr(), out.data_ptr(), B, H, W, Cin,
                    Cout, relu, cur_stream());
  return out;
}

torch::Tensor conv3x3_fp8(torch::Tensor x, torch::Tensor w_packed,
                          torch::Tensor bias, torch::Tensor w_scale,
                          torch::Tensor a_scale, bool relu) {
  // x: [B,Cin,H,W] channels_last, e4m3 passed as uint8 view;
  // w_packed: [9,Cout,Cin] uint8 (e4m3, tap-major); w_scale: [Cout]
  // f32; a_scale: [1] f32 DEVICE tensor (no host sync). Out: bf16 NHWC.
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "x must be [B,C,H,W] cuda");
  TORCH_CHECK(x.scalar_type() == torch::kUInt8,
              "x must be a uint8 view of e4m3");
  TORCH_CHECK(is_channels_last(x), "x must be channels_last (NHWC)");
  TORCH_CHECK(w_packed.dim() == 3 && w_packed.size(0) == 9 &&
                  w_packed.scalar_type() == torch::kUInt8 &&
                  w_packed.is_contiguous(),
              "w_packed must be contiguous uint8 [9, Cout, Cin]");
  const int B = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
  const int Cout = w_packed.size(1);
  TORCH_CHECK(w_packed.size(2) == Cin, "weight Cin mismatch");
  TORCH_CHECK(Cin % 32 == 0, "Cin must be a multiple of 32");
  TORCH_CHECK(a_scale.is_cuda() && a_scale.numel() == 1 &&
                  a_scale.scalar_type() == torch::kFloat,
              "a_scale must be a 1-elem f32 device tensor");
  auto bias_f = bias.to(torch::kFloat).contiguous();
  auto wsc = w_scale.to(torch::kFloat).contiguous();
  TORCH_CHECK(wsc.numel() == Cout, "w_scale must be [Cout]");
  auto out = torch::empty(
      {B, Cout, H, W},
      x.options().dtype(torch::kBFloat16)
          .memory_format(at::MemoryFormat::ChannelsLast));
  conv3x3_fp8_launch(x.data_ptr(), w_packed.data_ptr(),
                     bias_f.data_ptr<float>(), wsc.data_ptr<float>(),
                     a_scale.data_ptr<float>(), out.data_ptr(),
                     B, H, W, Cin, Cout, relu, cur_stream());
  return out;
}

torch::Tensor conv3x3_wrw(torch::Tensor x, torch::Tensor dy) {
  // x: [B,Cin,H,W], dy: [B,Cout,H,W] — both channels_last bf16.
  // Returns dW fp32 as [9, Cout, Cin] (tap-major; the python wrapper
  // reshapes to the torch [Cout, Cin, 3, 3] layout).
  TORCH_CHECK(x.is_cuda() && dy.is_cuda(), "conv3x3_wrw: GPU tensors");
  TORCH_CHECK(x.dim() == 4 && dy.dim() == 4, "x/dy must be 4-D");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
                  dy.scalar_type() == torch::kBFloat16,
              "conv3x3_wrw: bf16 only");
  TORCH_CHECK(is_channels_last(x) && is_channels_last(dy),
              "conv3x3_wrw: channels_last required");
  const int B = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
  const int Cout = dy.size(1);
  TORCH_CHECK(dy.size(0) == B && dy.size(2) == H && dy.size(3) == W,
              "conv3x3_wrw: x/dy spatial mismatch");
  TORCH_CHECK(Cin % 8 == 0 && Cout % 8 == 0,
              "conv3x3_wrw: channels must be multiples of 8");
  auto dw = torch::zeros({9, Cout, Cin},
                         x.options().dtype(torch::kFloat));
  conv3x3_wrw_launch(x.data_ptr(), dy.data_ptr(), dw.data_ptr<float>(),
                     B, H, W, Cin, Cout, cur_stream());
  return dw;
}

std::vector<torch::Tensor> detection_prep(
    torch::Tensor rois, torch::Tensor probs, torch::Tensor deltas,
    torch::Tensor windows, torch::Tensor std_dev, double min_conf) {
  // rois [B,N,4], probs [B,N,C], deltas [B,N,C,4] (fp32 or bf16,
  // same dtype); windows [B,4] f32; std_dev [4] f32.
  TORCH_CHECK(rois.is_cuda() && probs.is_cuda() && deltas.is_cuda(),
              "detection_prep: GPU tensors");
  TORCH_CHECK(rois.dim() == 3 && probs.dim() == 3 && deltas.dim() == 4,
              "detection_prep: bad ranks");
  const int B = rois.size(0), N = rois.size(1), C = probs.size(2);
  TORCH_CHECK(deltas.size(2) == C && deltas.size(3) == 4,
              "deltas must be [B,N,C,4]");
  auto dtype = probs.scalar_type();
  TORCH_CHECK(dtype == torch::kFloat || dtype == torch::kBFloat16,
              "detection_prep: fp32/bf16 only");
  auto r = rois.to(torch::kFloat).contiguous();  // never downcast boxes
  auto p = probs.contiguous();
  auto d = deltas.to(dtype).contiguous();
  auto w = windows.to(torch::kFloat).contiguous();
  auto sd = std_dev.to(torch::kFloat).contiguous();
  TORCH_CHECK(w.is_cuda() && w.numel() == (long)B * 4, "windows [B,4]");
  TORCH_CHECK(sd.is_cuda() && sd.numel() == 4, "std_dev [4]");
  auto refined = torch::empty({B, N, 4},
                              r.options().dtype(torch::kFloat));
  auto cls = torch::empty({B, N}, r.options().dtype(torch::kInt));
  auto scores = torch::empty({B, N}, r.options().dtype(torch::kFloat));
  detection_prep_launch(r.data_ptr(), p.data_ptr(), d.data_ptr(),
                        w.data_ptr<float>(), sd.data_ptr<float>(),
                        refined.data_ptr<float>(), cls.data_ptr<int>(),
                        scores.data_ptr<float>(), B, N, C,
                        (float)min_conf, dtype == torch::kBFloat16,
                        cur_stream());
  return {refined, cls, scores};
}

}  // namespace

void register_rle(pybind11::module_& m);
void register_unmold(pybind11::module_& m);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("roi_align_forward", &roi_align_forward,
        "RoIAlign / crop_and_resize forward (CDNA4)");
  m.def("roi_align_backward", &roi_align_backward,
        "RoIAlign backward (CDNA4, atomic scatter)");
  m.def("roi_align_backward_det", &roi_align_backward_det,
        "RoIAlign backward, deterministic fixed-point i64 reduction");
  m.def("nms", &nms, "bitmask greedy NMS (CDNA4)");
  m.def("nms_padded", &nms_padded,
        "sync-free NMS: (-1-padded keep [max_out], device count)",
        py::arg("boxes"), py::arg("scores"), py::arg("iou_thresh"),
        py::arg("max_out"));
  m.def("fpn_fuse", &fpn_fuse,
        "fused 1x1 conv + bias + nearest-up2 add (CDNA4 MFMA)",
        py::arg("x"), py::arg("w"), py::arg("bias"),
        py::arg("up") = py::none());
  register_rle(m);
  register_unmold(m);
  m.def("conv3x3_fwd", &conv3x3_fwd,
        "3x3 conv s1p1 + bias + optional ReLU (CDNA4 MFMA implicit GEMM)",
        py::arg("x"), py::arg("w_packed"), py::arg("bias"),
        py::arg("relu") = false, py::arg("version") = 0);
  m.def("conv3x3_fp8", &conv3x3_fp8,
        "3x3 conv s1p1, e4m3 in / bf16 out (fp8 MFMA rate)",
        py::arg("x"), py::arg("w_packed"), py::arg("bias"),
        py::arg("w_scale"), py::arg("a_scale"),
        py::arg("relu") = false);
  m.def("detection_prep", &detection_prep,
        "fused argmax/class-delta/clip detection prep (N9)",
        py::arg("rois"), py::arg("probs"), py::arg("deltas"),
        py::arg("windows"), py::arg("std_dev"), py::arg("min_conf"));
  m.def("deconv2x2", &deconv2x2,
        "2x2 stride-2 transposed conv + bias (+ReLU), MFMA (CDNA4)");
  m.def("conv3x3_mx", &conv3x3_mx,
        "MX block-scaled e4m3 3x3 conv (mfma_scale 16x16x128, gfx950)");
  m.def("mx_quant", &mx_quant,
        "fused MX block-64 activation quantization (bf16 NHWC -> e4m3)");
  m.def("mx_probe", &mx_probe,
        "one-instruction mfma_scale layout probe (test-only)");
  m.def("conv3x3_wrw", &conv3x3_wrw,
        "3x3 conv s1p1 weight gradient (CDNA4 MFMA, split-K atomics)",
        py::arg("x"), py::arg("dy"));
}
