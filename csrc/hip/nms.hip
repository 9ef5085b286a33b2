// Bitmask greedy NMS — CDNA4 (gfx950).
//
// Replaces tf.image.non_max_suppression for ProposalLayer
// (6000 -> 2000/1000 rois) and DetectionLayer (per-class via the
// coordinate-offset trick, one launch for all classes).
//
// Two-phase: (1) a parallel triangular IoU pass builds a [N, ceil(N/64)]
// u64 suppression matrix, 64 candidate boxes staged in LDS per block and
// one wave64 per row-box group (wavefront-sized bitmasks, not warp-32);
// (2) a single-workgroup serial scan resolves the greedy chain on-GPU
// (no DtoH of the matrix, no host sync until the caller reads the count).
// Boxes must be pre-sorted by descending score (the binding sorts).
#include "common.h"

namespace {

constexpr int BOX_BLOCK = 64;  // one wavefront of boxes per mask word

__device__ inline float iou_yx(const float4 a, const float4 b) {
  // boxes as (y1, x1, y2, x2)
  float yy1 = fmaxf(a.x, b.x);
  float xx1 = fmaxf(a.y, b.y);
  float yy2 = fminf(a.z, b.z);
  float xx2 = fminf(a.w, b.w);
  float ih = fmaxf(yy2 - yy1, 0.f);
  float iw = fmaxf(xx2 - xx1, 0.f);
  float inter = ih * iw;
  float areaA = fmaxf(a.z - a.x, 0.f) * fmaxf(a.w - a.y, 0.f);
  float areaB = fmaxf(b.z - b.x, 0.f) * fmaxf(b.w - b.y, 0.f);
  float uni = areaA + areaB - inter;
  return uni > 0.f ? inter / uni : 0.f;
}

// Grid: (ceil(N/64) row blocks, ceil(N/64) col blocks); 64 threads each.
__global__ void nms_mask_kernel(const float* __restrict__ boxes, int N,
                                float thresh, unsigned long long* mask,
                                int nblocks) {
  const int row_block = blockIdx.x;
  const int col_block = blockIdx.y;
  if (col_block < row_block) return;

  __shared__ float4 col_boxes[BOX_BLOCK];
  const int col_base = col_block * BOX_BLOCK;
  const int row_base = row_block * BOX_BLOCK;
  const int tid = threadIdx.x;

  if (col_base + tid < N) {
    const float* b = boxes + (col_base + tid) * 4;
    col_boxes[tid] = make_float4(b[0], b[1], b[2], b[3]);
  }
  __syncthreads();

  const int i = row_base + tid;
  if (i >= N) return;
  const float* bp = boxes + i * 4;
  const float4 bi = make_float4(bp[0], bp[1], bp[2], bp[3]);

  unsigned long long bits = 0ull;
  const int n_cols = min(BOX_BLOCK, N - col_base);
  // Only j > i suppressions matter (boxes are score-sorted).
  int j_start = (row_block == col_block) ? tid + 1 : 0;
  for (int j = j_start; j < n_cols; ++j) {
    if (iou_yx(bi, col_boxes[j]) > thresh) bits |= (1ull << j);
  }
  mask[(long)i * nblocks + col_block] = bits;
}

// Single workgroup resolves the greedy chain. keep_out gets the kept
// SORTED indices; count_out the number kept.
__global__ void nms_scan_kernel(const unsigned long long* __restrict__ mask,
                                int N, int nblocks, int max_out,
                                int* __restrict__ keep_out,
                                int* __restrict__ count_out) {
  extern __shared__ unsigned long long alive[];  // nblocks words
  const int tid = threadIdx.x;
  for (int w = tid; w < nblocks; w += blockDim.x) {
    int valid = N - w * BOX_BLOCK;
    alive[w] = (valid >= BOX_BLOCK) ? ~0ull
               : ((valid <= 0) ? 0ull : ((1ull << valid) - 1ull));
  }
  __syncthreads();

  __shared__ int count;
  if (tid == 0) count = 0;
  __syncthreads();

  for (int i = 0; i < N; ++i) {
    const int w = i / BOX_BLOCK, b = i % BOX_BLOCK;
    // All threads read the same word: LDS broadcast, no conflict.
    bool is_alive = (alive[w] >> b) & 1ull;
    if (is_alive) {
      if (tid == 0) {
        keep_out[count] = i;
        ++count;
      }
      __syncthreads();
      if (count >= max_out) break;
      // Clear the boxes i suppresses.
      const unsigned long long* row = mask + (long)i * nblocks;
      for (int ww = tid; ww < nblocks; ww += blockDim.x) {
        alive[ww] &= ~row[ww];
      }
      __syncthreads();
    }
  }
  if (tid == 0) *count_out = min(count, max_out);
}

}  // namespace

extern "C" {

// workspace layout: mask (N*nblocks u64). keep_out int32 [max_out],
// count_out int32 [1].
long nms_workspace_bytes(int N) {
  int nblocks = ceil_div(N, BOX_BLOCK);
  return (long)N * nblocks * sizeof(unsigned long long);
}

void nms_launch(const float* boxes_sorted, int N, float iou_thresh,
                int max_out, void* workspace, int* keep_out, int* count_out,
                hipStream_t stream) {
  if (N == 0) {
    hipMemsetAsync(count_out, 0, sizeof(int), stream);
    return;
  }
  int nblocks = ceil_div(N, BOX_BLOCK);
  unsigned long long* mask = (unsigned long long*)workspace;
  dim3 grid(nblocks, nblocks);
  hipLaunchKernelGGL(nms_mask_kernel, grid, dim3(BOX_BLOCK), 0, stream,
                     boxes_sorted, N, iou_thresh, mask, nblocks);
  hipLaunchKernelGGL(nms_scan_kernel, dim3(1), dim3(256),
                     nblocks * sizeof(unsigned long long), stream, mask, N,
                     nblocks, max_out, keep_out, count_out);
}

}  // extern "C"
