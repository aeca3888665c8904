// Fused causal-LM cross-entropy for gfx950 (SURVEY.md §2b "Causal-LM
// cross-entropy"; reference computes outputs.loss via transformers, which
// upcasts the full [B,S,V] logits to fp32 — the single biggest activation.
// This kernel never materializes an fp32 copy: one online-logsumexp pass over
// the bf16 logits per row, f32 accumulation, loss and lse saved per row).
//
// Shift semantics (labels[t+1] predicts from logits[t]) are handled by the
// caller via row indexing: loss rows r in [0, B*(S-1)), logits row offset
// b*S + s with b = r/(S-1), s = r%(S-1), label = labels[b*S + s + 1].
// The kernel takes a generic row->(logit_row, label) mapping through
// S_logits/S_out so no contiguous copy of the sliced logits is ever made.
//
// Vocab-sharded variant (loss parallel, 06-tensor-parallel/README.md:243-271):
// ce_sharded_* computes local (max, sumexp, gathered-logit) per row; the
// Python side all-reduces the three scalars per token over the TP group and
// finishes the loss — see ops/cross_entropy.py.
#include "common.h"

// one block per row, online max/sum in a single pass
template <bool SHARDED>
__global__ void __launch_bounds__(256) ce_fwd_kernel(
    const short* __restrict__ logits, const int64_t* __restrict__ labels,
    float* __restrict__ loss, float* __restrict__ lse,
    float* __restrict__ maxout,      // SHARDED only: local max per row
    float* __restrict__ sumout,      // SHARDED only: local sum(exp(x-max))
    float* __restrict__ gathered,    // SHARDED only: x[label] or -inf
    int64_t nrows, int S_logits, int S_out, int V,
    int64_t vocab_start, int64_t ignore_index) {
  __shared__ float scratch[8];
  for (int64_t r = blockIdx.x; r < nrows; r += gridDim.x) {
    int64_t b = r / S_out;
    int64_t s = r % S_out;
    const short* xr = logits + (b * S_logits + s) * V;
    int64_t label = labels[b * S_logits + s + (S_logits - S_out)];
    // online max/sum over V
    float m = -INFINITY, sum = 0.0f;
    for (int i = threadIdx.x * 8; i < V; i += blockDim.x * 8) {
      s16x8 v;
      if (i + 8 <= V) v = *reinterpret_cast<const s16x8*>(xr + i);
      else { for (int j = 0; j < 8; ++j) v[j] = (i + j < V) ? xr[i + j] : (short)0xff80; /* -inf bf16 */ }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(v[j]);
        if (f > m) { sum *= __expf(m - f); m = f; }
        sum += __expf(f - m);
      }
    }
    // block combine: max then rescaled sums
    float gm = block_reduce_max(m, scratch);
    sum = (m == -INFINITY) ? 0.0f : sum * __expf(m - gm);
    sum = block_reduce_sum(sum, scratch);
    if (threadIdx.x == 0) {
      if (SHARDED) {
        float g = -INFINITY;
        int64_t local = label - vocab_start;
        if (label != ignore_index && local >= 0 && local < V)
          g = bf2f(xr[local]);
        maxout[r] = gm;
        sumout[r] = sum;
        gathered[r] = g;
      } else {
        if (label == ignore_index) {
          loss[r] = 0.0f;
          lse[r] = -INFINITY;  // marks "ignored" for the backward
        } else {
          DTGA_KERNEL_ASSERT(label >= 0 && label < V);
          float l = gm + __logf(sum);
          lse[r] = l;
          loss[r] = l - bf2f(xr[label]);
        }
      }
    }
    __syncthreads();
  }
}

// dlogits[r, v] = scale_r * (softmax - onehot); scale_r = dloss * row_weight
// row_weight handled by caller through `scale` (e.g. 1/n_valid for mean).
template <bool SHARDED>
__global__ void __launch_bounds__(256) ce_bwd_kernel(
    const short* __restrict__ logits, const int64_t* __restrict__ labels,
    const float* __restrict__ lse, short* __restrict__ dlogits, float scale,
    const float* __restrict__ scale_ptr,  // device scalar (no host sync)
    int64_t nrows, int S_logits, int S_out, int V,
    int64_t vocab_start, int64_t ignore_index) {
  if (scale_ptr != nullptr) scale = scale_ptr[0];
  for (int64_t r = blockIdx.x; r < nrows; r += gridDim.x) {
    int64_t b = r / S_out;
    int64_t s = r % S_out;
    const short* xr = logits + (b * S_logits + s) * V;
    short* dxr = dlogits + (b * S_logits + s) * V;
    int64_t label = labels[b * S_logits + s + (S_logits - S_out)];
    float l = lse[r];
    bool ignored = (label == ignore_index) || (l == -INFINITY);
    if (!SHARDED) DTGA_KERNEL_ASSERT(ignored || (label >= 0 && label < V));
    int64_t local = label - vocab_start;
    for (int i = threadIdx.x * 8; i < V; i += blockDim.x * 8) {
      if (i + 8 <= V) {
        s16x8 v = *reinterpret_cast<const s16x8*>(xr + i);
        s16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float p = ignored ? 0.0f : __expf(bf2f(v[j]) - l);
          if (!ignored && (int64_t)(i + j) == local) p -= 1.0f;
          o[j] = f2bf(scale * p);
        }
        *reinterpret_cast<s16x8*>(dxr + i) = o;
      } else {
        for (int j = 0; i + j < V; ++j) {
          float p = ignored ? 0.0f : __expf(bf2f(xr[i + j]) - l);
          if (!ignored && (int64_t)(i + j) == local) p -= 1.0f;
          dxr[i + j] = f2bf(scale * p);
        }
      }
    }
    // rows of dlogits for the last position (never a loss row) are zeroed by
    // the caller once (memset) — this kernel only touches loss rows.
  }
}

extern "C" {
void ce_fwd_launch(const void* logits, const int64_t* labels, float* loss,
                   float* lse, int64_t nrows, int S_logits, int S_out, int V,
                   int64_t ignore_index, hipStream_t st) {
  int grid = (int)(nrows < 2048 ? (nrows < 1 ? 1 : nrows) : 2048);
  hipLaunchKernelGGL((ce_fwd_kernel<false>), dim3(grid), dim3(256), 0, st,
                     (const short*)logits, labels, loss, lse, nullptr, nullptr,
                     nullptr, nrows, S_logits, S_out, V, 0, ignore_index);
}
void ce_fwd_sharded_launch(const void* logits, const int64_t* labels,
                           float* maxout, float* sumout, float* gathered,
                           int64_t nrows, int S_logits, int S_out, int V,
                           int64_t vocab_start, int64_t ignore_index,
                           hipStream_t st) {
  int grid = (int)(nrows < 2048 ? (nrows < 1 ? 1 : nrows) : 2048);
  hipLaunchKernelGGL((ce_fwd_kernel<true>), dim3(grid), dim3(256), 0, st,
                     (const short*)logits, labels, nullptr, nullptr, maxout,
                     sumout, gathered, nrows, S_logits, S_out, V, vocab_start,
                     ignore_index);
}
void ce_bwd_launch(const void* logits, const int64_t* labels, const float* lse,
                   void* dlogits, float scale, const float* scale_ptr,
                   int64_t nrows, int S_logits, int S_out, int V,
                   int64_t vocab_start, int64_t ignore_index, int sharded,
                   hipStream_t st) {
  int grid = (int)(nrows < 2048 ? (nrows < 1 ? 1 : nrows) : 2048);
  if (sharded)
    hipLaunchKernelGGL((ce_bwd_kernel<true>), dim3(grid), dim3(256), 0, st,
                       (const short*)logits, labels, lse, (short*)dlogits,
                       scale, scale_ptr, nrows, S_logits, S_out, V,
                       vocab_start, ignore_index);
  else
    hipLaunchKernelGGL((ce_bwd_kernel<false>), dim3(grid), dim3(256), 0, st,
                       (const short*)logits, labels, lse, (short*)dlogits,
                       scale, scale_ptr, nrows, S_logits, S_out, V,
                       vocab_start, ignore_index);
}
}
