// Elementwise / row-wise kernels: RMSNorm(+residual), SwiGLU, RoPE,
// row softmax, masked mean-pool + L2-norm, sampling.
// All bf16 I/O with fp32 accumulation; vectorized bf16x8 loads per
// cdna_hip_programming.md G13 (scalar bf16 loads are ~2x slower).
#include "common.h"

// ---------------------------------------------------------------- rmsnorm
// x [N, D] bf16, w [D] bf16 -> y [N, D] bf16.  One block per row.
// Two passes over the row: sumsq (f32), then normalize; second read hits L1/L2.
template <int VEC>
__global__ void rmsnorm_kernel(const bf16* __restrict__ x,
                               const bf16* __restrict__ w,
                               bf16* __restrict__ y, int D, float eps) {
  __shared__ float scratch[32];
  const long row = blockIdx.x;
  const bf16* xr = x + row * (long)D;
  bf16* yr = y + row * (long)D;

  float ss = 0.f;
  for (int d = threadIdx.x * VEC; d < D; d += blockDim.x * VEC) {
    if (VEC == 8) {
      bf16x8 v = *reinterpret_cast<const bf16x8*>(xr + d);
#pragma unroll
      for (int j = 0; j < 8; ++j) { float f = bits2f(v[j]); ss += f * f; }
    } else {
      float f = bf2f(xr[d]);
      ss += f * f;
    }
  }
  ss = block_sum(ss, scratch);
  const float inv = rsqrtf(ss / (float)D + eps);

  for (int d = threadIdx.x * VEC; d < D; d += blockDim.x * VEC) {
    if (VEC == 8) {
      bf16x8 v = *reinterpret_cast<const bf16x8*>(xr + d);
      bf16x8 wv = *reinterpret_cast<const bf16x8*>(w + d);
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = f2bits(bits2f(v[j]) * inv * bits2f(wv[j]));
      *reinterpret_cast<bf16x8*>(yr + d) = o;
    } else {
      yr[d] = f2bf(bf2f(xr[d]) * inv * bf2f(w[d]));
    }
  }
}

// h = x + res; y = rmsnorm(h) * w.  Writes both h and y (fused residual).
template <int VEC>
__global__ void rmsnorm_residual_kernel(const bf16* __restrict__ x,
                                        const bf16* __restrict__ res,
                                        const bf16* __restrict__ w,
                                        bf16* __restrict__ y,
                                        bf16* __restrict__ h,
                                        int D, float eps) {
  __shared__ float scratch[32];
  const long row = blockIdx.x;
  const bf16* xr = x + row * (long)D;
  const bf16* rr = res + row * (long)D;
  bf16* yr = y + row * (long)D;
  bf16* hr = h + row * (long)D;

  float ss = 0.f;
  for (int d = threadIdx.x * VEC; d < D; d += blockDim.x * VEC) {
    if (VEC == 8) {
      bf16x8 a = *reinterpret_cast<const bf16x8*>(xr + d);
      bf16x8 b = *reinterpret_cast<const bf16x8*>(rr + d);
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bits2f(a[j]) + bits2f(b[j]);
        o[j] = f2bits(f);
        ss += f * f;
      }
      *reinterpret_cast<bf16x8*>(hr + d) = o;
    } else {
      float f = bf2f(xr[d]) + bf2f(rr[d]);
      hr[d] = f2bf(f);
      ss += f * f;
    }
  }
  ss = block_sum(ss, scratch);
  const float inv = rsqrtf(ss / (float)D + eps);
  for (int d = threadIdx.x * VEC; d < D; d += blockDim.x * VEC) {
    if (VEC == 8) {
      bf16x8 v = *reinterpret_cast<const bf16x8*>(hr + d);
      bf16x8 wv = *reinterpret_cast<const bf16x8*>(w + d);
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = f2bits(bits2f(v[j]) * inv * bits2f(wv[j]));
      *reinterpret_cast<bf16x8*>(yr + d) = o;
    } else {
      yr[d] = f2bf(bf2f(hr[d]) * inv * bf2f(w[d]));
    }
  }
}

// ---------------------------------------------------------------- swiglu
// out = silu(gate) * up, flat N elements, grid-stride, bf16x8 vectorized.
__global__ void swiglu_kernel(const bf16* __restrict__ gate,
                              const bf16* __restrict__ up,
                              bf16* __restrict__ out, long n8) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    bf16x8 g = reinterpret_cast<const bf16x8*>(gate)[i];
    bf16x8 u = reinterpret_cast<const bf16x8*>(up)[i];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bits2f(g[j]);
      float s = gf / (1.f + __expf(-gf));
      o[j] = f2bits(s * bits2f(u[j]));
    }
    reinterpret_cast<bf16x8*>(out)[i] = o;
  }
}
// Packed variant: gu [N, 2F] rows = [gate | up]; out [N, F].
// Avoids the two strided .contiguous() copies per FFN invocation.
__global__ void swiglu_packed_kernel(const bf16* __restrict__ gu,
                                     bf16* __restrict__ out, long rows, int F) {
  const int f8 = F / 8;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x;
       i < rows * (long)f8; i += (long)gridDim.x * blockDim.x) {
    const long r = i / f8;
    const int c8 = (int)(i % f8);
    const bf16* row = gu + r * (long)(2 * F);
    bf16x8 g = *reinterpret_cast<const bf16x8*>(row + c8 * 8);
    bf16x8 u = *reinterpret_cast<const bf16x8*>(row + F + c8 * 8);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bits2f(g[j]);
      float s = gf / (1.f + __expf(-gf));
      o[j] = f2bits(s * bits2f(u[j]));
    }
    *reinterpret_cast<bf16x8*>(out + r * (long)F + c8 * 8) = o;
  }
}

__global__ void swiglu_tail_kernel(const bf16* __restrict__ gate,
                                   const bf16* __restrict__ up,
                                   bf16* __restrict__ out, long start, long n) {
  long i = start + blockIdx.x * (long)blockDim.x + threadIdx.x;
  if (i < n) {
    float gf = bf2f(gate[i]);
    float s = gf / (1.f + __expf(-gf));
    out[i] = f2bf(s * bf2f(up[i]));
  }
}

// ---------------------------------------------------------------- rope
// x [B, S, H, D] bf16 in-place-able; cos/sin [maxS, D/2] f32; pos [B, S] i32.
// Each thread rotates one (pair) element: total B*S*H*D/2 pairs.
__global__ void rope_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
                            const float* __restrict__ cosT,
                            const float* __restrict__ sinT,
                            const int* __restrict__ pos,
                            int S, int H, int D, long total_pairs) {
  const int half = D / 2;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total_pairs;
       i += (long)gridDim.x * blockDim.x) {
    long pair = i;                 // ((b*S + s)*H + h)*half + p
    int p = (int)(pair % half);
    long rest = pair / half;       // (b*S + s)*H + h
    long bs = rest / H;            // b*S + s
    int pp = pos[bs];
    float c = cosT[(long)pp * half + p];
    float sn = sinT[(long)pp * half + p];
    long base = rest * (long)D + 2 * p;
    float x0 = bf2f(x[base]);
    float x1 = bf2f(x[base + 1]);
    y[base] = f2bf(x0 * c - x1 * sn);
    y[base + 1] = f2bf(x0 * sn + x1 * c);
  }
}

// ------------------------------------------------------- decode qkv prep
// Fused per-token decode head prep: input is the raw QKV projection
// [B, (H+2*Hkv)*D] for S=1; this kernel applies RoPE to the q and k heads
// and scatters k/v into the KV caches at row seq_lens[b] — replacing a
// rope launch for q, a rope launch for k, and two indexed cache-write
// launches per layer per token (the decode step's tiny-kernel tail).
// Grid: (H + 2*Hkv, B); block: 128 threads.
__global__ void decode_qkv_prep_kernel(
    const bf16* __restrict__ qkv, bf16* __restrict__ q_out,
    bf16* __restrict__ kc, bf16* __restrict__ vc,
    const float* __restrict__ cosT, const float* __restrict__ sinT,
    const int* __restrict__ seq_lens, int H, int Hkv, int D, int Smax) {
  const int b = blockIdx.y;
  const int h = blockIdx.x;            // 0..H+2*Hkv-1
  const int pos = seq_lens[b];
  const int half = D / 2;
  const bf16* src = qkv + ((long)b * (H + 2 * Hkv) + h) * D;

  if (h < H + Hkv) {                   // q or k head: rotate
    bf16* dst = (h < H)
        ? q_out + ((long)b * H + h) * D
        : kc + (((long)b * Hkv + (h - H)) * Smax + pos) * (long)D;
    for (int p = threadIdx.x; p < half; p += blockDim.x) {
      const float c = cosT[(long)pos * half + p];
      const float sn = sinT[(long)pos * half + p];
      const float x0 = bf2f(src[2 * p]);
      const float x1 = bf2f(src[2 * p + 1]);
      dst[2 * p] = f2bf(x0 * c - x1 * sn);
      dst[2 * p + 1] = f2bf(x0 * sn + x1 * c);
    }
  } else {                             // v head: copy into cache
    bf16* dst = vc + (((long)b * Hkv + (h - H - Hkv)) * Smax + pos) * (long)D;
    for (int d = threadIdx.x; d < D; d += blockDim.x) dst[d] = src[d];
  }
}

// ---------------------------------------------------------------- softmax
// x [N, D] bf16 -> y [N, D] bf16, row-wise, online in two block passes.
__global__ void softmax_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
                               int D) {
  __shared__ float scratch[32];
  const long row = blockIdx.x;
  const bf16* xr = x + row * (long)D;
  bf16* yr = y + row * (long)D;

  float m = -INFINITY;
  for (int d = threadIdx.x; d < D; d += blockDim.x) m = fmaxf(m, bf2f(xr[d]));
  m = block_max(m, scratch);
  float s = 0.f;
  for (int d = threadIdx.x; d < D; d += blockDim.x) s += __expf(bf2f(xr[d]) - m);
  s = block_sum(s, scratch);
  const float inv = 1.f / s;
  for (int d = threadIdx.x; d < D; d += blockDim.x)
    yr[d] = f2bf(__expf(bf2f(xr[d]) - m) * inv);
}

// ------------------------------------------------- mean-pool + L2 normalize
// hidden [B, S, D] bf16, mask [B, S] u8 -> out [B, D] f32.  Block per batch.
__global__ void mean_pool_l2norm_kernel(const bf16* __restrict__ hidden,
                                        const unsigned char* __restrict__ mask,
                                        float* __restrict__ out,
                                        int S, int D) {
  __shared__ float scratch[32];
  const int b = blockIdx.x;
  const bf16* hb = hidden + (long)b * S * D;
  float* ob = out + (long)b * D;

  float cnt = 0.f;
  for (int s = threadIdx.x; s < S; s += blockDim.x)
    cnt += mask[(long)b * S + s] ? 1.f : 0.f;
  cnt = block_sum(cnt, scratch);
  const float invc = 1.f / fmaxf(cnt, 1.f);

  // each thread owns d-elements stride blockDim
  float normsq = 0.f;
  for (int d = threadIdx.x; d < D; d += blockDim.x) {
    float acc = 0.f;
    for (int s = 0; s < S; ++s) {
      if (mask[(long)b * S + s])
        acc += bf2f(hb[(long)s * D + d]);
    }
    acc *= invc;
    ob[d] = acc;                  // staged un-normalized
    normsq += acc * acc;
  }
  normsq = block_sum(normsq, scratch);
  const float inv = rsqrtf(fmaxf(normsq, 1e-24f));
  for (int d = threadIdx.x; d < D; d += blockDim.x) ob[d] *= inv;
}

// ---------------------------------------------------------------- sampling
// logits [B, V] f32 -> token [B] i64.  Gumbel-argmax at T>0 (equivalent to
// softmax sampling, single pass, no normalization); plain argmax at T<=0.
// Two phases: the Gumbel-perturbed scan is transcendental-bound (2 logs per
// element), so each row splits across SAMPLE_PARTS blocks; a tiny combine
// kernel reduces the per-part winners.
#define SAMPLE_PARTS 16
__global__ void sample_part_kernel(const float* __restrict__ logits,
                                   float* __restrict__ ws_val,
                                   int* __restrict__ ws_idx, int V,
                                   float invT, unsigned seed, int greedy) {
  __shared__ float s_val[32];
  __shared__ int s_idx[32];
  const int b = blockIdx.x;
  const int part = blockIdx.y;
  const float* lb = logits + (long)b * V;
  const int span = (V + SAMPLE_PARTS - 1) / SAMPLE_PARTS;
  const int v_lo = part * span;
  const int v_hi = min(V, v_lo + span);

  float best = -INFINITY;
  int besti = v_lo;
  for (int v = v_lo + threadIdx.x; v < v_hi; v += blockDim.x) {
    float sc = lb[v];
    if (!greedy) {
      unsigned h = hash_u32(seed, (unsigned)b, (unsigned)v);
      float u = ((float)h + 1.0f) * (1.0f / 4294967808.0f);  // (0,1)
      float g = -__logf(-__logf(u));
      sc = sc * invT + g;
    }
    if (sc > best) { best = sc; besti = v; }
  }
  // wave then block argmax
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ov = __shfl_xor(best, off, WAVE);
    int oi = __shfl_xor(besti, off, WAVE);
    if (ov > best || (ov == best && oi < besti)) { best = ov; besti = oi; }
  }
  if (lane == 0) { s_val[wid] = best; s_idx[wid] = besti; }
  __syncthreads();
  if (threadIdx.x == 0) {
    int nw = blockDim.x / WAVE;
    float bv = s_val[0]; int bi = s_idx[0];
    for (int w = 1; w < nw; ++w)
      if (s_val[w] > bv || (s_val[w] == bv && s_idx[w] < bi)) {
        bv = s_val[w]; bi = s_idx[w];
      }
    ws_val[b * SAMPLE_PARTS + part] = bv;
    ws_idx[b * SAMPLE_PARTS + part] = bi;
  }
}

__global__ void sample_combine_kernel(const float* __restrict__ ws_val,
                                      const int* __restrict__ ws_idx,
                                      long* __restrict__ out) {
  const int b = blockIdx.x;
  float bv = -INFINITY;
  int bi = 0;
  for (int p = 0; p < SAMPLE_PARTS; ++p) {
    const float v = ws_val[b * SAMPLE_PARTS + p];
    const int i = ws_idx[b * SAMPLE_PARTS + p];
    if (v > bv || (v == bv && i < bi)) { bv = v; bi = i; }
  }
  out[b] = bi;
}

// ---------------------------------------------------------------- C API
extern "C" {

hipError_t sentio_rmsnorm(const void* x, const void* w, void* y, long rows,
                          int D, float eps, hipStream_t stream) {
  dim3 block(256);
  if (D % 8 == 0)
    hipLaunchKernelGGL((rmsnorm_kernel<8>), dim3((unsigned)rows), block, 0,
                       stream, (const bf16*)x, (const bf16*)w, (bf16*)y, D, eps);
  else
    hipLaunchKernelGGL((rmsnorm_kernel<1>), dim3((unsigned)rows), block, 0,
                       stream, (const bf16*)x, (const bf16*)w, (bf16*)y, D, eps);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sentio_rmsnorm_residual(const void* x, const void* res, const void* w,
                                   void* y, void* h, long rows, int D,
                                   float eps, hipStream_t stream) {
  dim3 block(256);
  if (D % 8 == 0)
    hipLaunchKernelGGL((rmsnorm_residual_kernel<8>), dim3((unsigned)rows), block,
                       0, stream, (const bf16*)x, (const bf16*)res,
                       (const bf16*)w, (bf16*)y, (bf16*)h, D, eps);
  else
    hipLaunchKernelGGL((rmsnorm_residual_kernel<1>), dim3((unsigned)rows), block,
                       0, stream, (const bf16*)x, (const bf16*)res,
                       (const bf16*)w, (bf16*)y, (bf16*)h, D, eps);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sentio_swiglu(const void* gate, const void* up, void* out, long n,
                         hipStream_t stream) {
  long n8 = n / 8;
  if (n8 > 0) {
    long blocks = (n8 + 255) / 256;
    if (blocks > 2048) blocks = 2048;  // grid-stride (guide G11)
    hipLaunchKernelGGL(swiglu_kernel, dim3((unsigned)blocks), dim3(256), 0,
                       stream, (const bf16*)gate, (const bf16*)up, (bf16*)out, n8);
  }
  long tail = n - n8 * 8;
  if (tail > 0) {
    hipLaunchKernelGGL(swiglu_tail_kernel, dim3(1), dim3(256), 0, stream,
                       (const bf16*)gate, (const bf16*)up, (bf16*)out,
                       n8 * 8, n);
  }
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sentio_rope(const void* x, void* y, const float* cosT,
                       const float* sinT, const int* pos, int B, int S, int H,
                       int D, hipStream_t stream) {
  long total = (long)B * S * H * (D / 2);
  long blocks = (total + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  hipLaunchKernelGGL(rope_kernel, dim3((unsigned)blocks), dim3(256), 0, stream,
                     (const bf16*)x, (bf16*)y, cosT, sinT, pos, S, H, D, total);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sentio_swiglu_packed(const void* gu, void* out, long rows, int F,
                                hipStream_t stream) {
  if (F % 8) return hipErrorInvalidValue;
  long total = rows * (long)(F / 8);
  long blocks = (total + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  hipLaunchKernelGGL(swiglu_packed_kernel, dim3((unsigned)blocks), dim3(256),
                     0, stream, (const bf16*)gu, (bf16*)out, rows, F);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sentio_decode_qkv_prep(const void* qkv, void* q_out, void* kc,
                                  void* vc, const float* cosT,
                                  const float* sinT, const int* seq_lens,
                                  int B, int H, int Hkv, int D, int Smax,
                                  hipStream_t stream) {
  if (D % 2) return hipErrorInvalidValue;
  hipLaunchKernelGGL(decode_qkv_prep_kernel, dim3(H + 2 * Hkv, B), dim3(128),
                     0, stream, (const bf16*)qkv, (bf16*)q_out, (bf16*)kc,
                     (bf16*)vc, cosT, sinT, seq_lens, H, Hkv, D, Smax);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sentio_softmax(const void* x, void* y, long rows, int D,
                          hipStream_t stream) {
  hipLaunchKernelGGL(softmax_kernel, dim3((unsigned)rows), dim3(256), 0, stream,
                     (const bf16*)x, (bf16*)y, D);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sentio_mean_pool_l2norm(const void* hidden, const unsigned char* mask,
                                   float* out, int B, int S, int D,
                                   hipStream_t stream) {
  hipLaunchKernelGGL(mean_pool_l2norm_kernel, dim3(B), dim3(256), 0, stream,
                     (const bf16*)hidden, mask, out, S, D);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sentio_sample(const float* logits, long* out, float* ws_val,
                         int* ws_idx, int B, int V, float temperature,
                         unsigned seed, hipStream_t stream) {
  int greedy = temperature <= 0.f;
  float invT = greedy ? 1.f : 1.f / temperature;
  hipLaunchKernelGGL(sample_part_kernel, dim3(B, SAMPLE_PARTS), dim3(256), 0,
                     stream, logits, ws_val, ws_idx, V, invT, seed, greedy);
  HIP_CHECK_LAUNCH();
  hipLaunchKernelGGL(sample_combine_kernel, dim3(B), dim3(1), 0, stream,
                     ws_val, ws_idx, out);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

}  // extern "C"
