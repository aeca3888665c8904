// Rotary position embedding (RoPE) for gfx950 — rotate-half convention
// (Llama: pairs (i, i + D/2)).
//
// Replaces the reference's LlamaRotaryEmbedding application (SURVEY.md §2b
// "RoPE"; explicit position_ids per 06-tensor-parallel/train_llm.py:210-212).
// cos/sin tables are precomputed on the host side (f32 [S, D/2]) — on-device
// trig per element would turn this memory-bound op VALU-bound (guide App. B).
//
// Layout: x is [B, S, H, D] contiguous (BSHD — the attention kernel's native
// layout, no transposes anywhere on the hot path). Forward rotates by +theta,
// backward by -theta (the rotation's transpose), selected by `backward`.
// `positions` (optional, int32 [S]) supplies absolute positions when the
// sequence is sharded (sequence parallelism) or offset (resume mid-sequence).
#include "common.h"

template <bool BWD>
__global__ void __launch_bounds__(256) rope_kernel(
    const short* __restrict__ x, short* __restrict__ y,
    const float* __restrict__ cs,  // [max_pos, D/2] interleaved cos
    const float* __restrict__ sn,  // [max_pos, D/2] sin
    const int* __restrict__ positions,  // nullable, [S]
    int64_t total_rows,  // B*S*H
    int S, int H, int D) {
  const int halfD = D / 2;
  // one row (= one head at one position) per 64-lane wave when D/2 <= 64;
  // grid-stride over flat (row, d2) pairs otherwise.
  int64_t nelem = total_rows * halfD;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < nelem; idx += (int64_t)gridDim.x * blockDim.x) {
    int d2 = (int)(idx % halfD);
    int64_t row = idx / halfD;
    int s = (int)((row / H) % S);
    int pos = positions ? positions[s] : s;
    const float c = cs[(int64_t)pos * halfD + d2];
    const float sv = sn[(int64_t)pos * halfD + d2];
    const short* xr = x + row * D;
    short* yr = y + row * D;
    float x0 = bf2f(xr[d2]);
    float x1 = bf2f(xr[d2 + halfD]);
    if (BWD) {
      // inverse rotation: [c, s; -s, c]^T = [c, -s; s, c]
      yr[d2] = f2bf(x0 * c + x1 * sv);
      yr[d2 + halfD] = f2bf(-x0 * sv + x1 * c);
    } else {
      yr[d2] = f2bf(x0 * c - x1 * sv);
      yr[d2 + halfD] = f2bf(x0 * sv + x1 * c);
    }
  }
}

extern "C" {
void rope_launch(const void* x, void* y, const float* cs, const float* sn,
                 const int* positions, int64_t total_rows, int S, int H, int D,
                 int backward, hipStream_t stream) {
  int64_t nelem = total_rows * (D / 2);
  int64_t want = (nelem + 255) / 256;
  int grid = (int)(want < 2048 ? (want < 1 ? 1 : want) : 2048);
  if (backward)
    hipLaunchKernelGGL((rope_kernel<true>), dim3(grid), dim3(256), 0, stream,
                       (const short*)x, (short*)y, cs, sn, positions,
                       total_rows, S, H, D);
  else
    hipLaunchKernelGGL((rope_kernel<false>), dim3(grid), dim3(256), 0, stream,
                       (const short*)x, (short*)y, cs, sn, positions,
                       total_rows, S, H, D);
}
}
