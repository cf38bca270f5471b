// Retrieval kernels: fused cosine scoring scan over the in-HBM index (K2)
// and BM25 CSR-postings scoring (K3).  Memory-bandwidth-bound by design:
// the cosine scan reads each index row exactly once and amortizes it over
// all B queries of the batch (queries staged in LDS).
#include "common.h"

// mat [N, D] f16-or-bf16 row-major (L2-normalized rows), q [B, D] same dtype
// (normalized), scores [B, N] f32.
// Block = 4 waves; queries staged once in LDS; each wave scans rows
// grid-strided, lane l owns elements [l*E, l*E+E) of a row (E = D/64).
// Wave reduces B dots per row via shfl.
// E = D/64 is a template constant so the row cache rv[] stays in registers
// (runtime-indexed ext arrays spill to scratch — guide §5.4 rule 20).
template <typename T, int E>
__global__ void cosine_scores_kernel(const T* __restrict__ mat,
                                     const T* __restrict__ q,
                                     float* __restrict__ scores,
                                     long N, int B) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* qs = reinterpret_cast<float*>(smem);  // [B][D] f32
  constexpr int D = E * WAVE;

  // cooperative query staging
  for (int i = threadIdx.x; i < B * D; i += blockDim.x)
    qs[i] = (float)q[i];
  __syncthreads();

  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int waves_per_block = blockDim.x / WAVE;
  const long wave_global = (long)blockIdx.x * waves_per_block + wid;
  const long wave_count = (long)gridDim.x * waves_per_block;

  typedef T tvec8 __attribute__((ext_vector_type(8)));
  for (long row = wave_global; row < N; row += wave_count) {
    const T* r = mat + row * (long)D + lane * E;
    float rv[E];
    // vectorized 16-byte row loads (scalar f16/bf16 loads are ~2.5x slower,
    // guide G13); E >= 8 is the hot path (D >= 512)
    if constexpr (E % 8 == 0) {
#pragma unroll
      for (int e8 = 0; e8 < E / 8; ++e8) {
        tvec8 v = *reinterpret_cast<const tvec8*>(r + e8 * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) rv[e8 * 8 + j] = (float)v[j];
      }
    } else {
#pragma unroll
      for (int e = 0; e < E; ++e) rv[e] = (float)r[e];
    }
    for (int b = 0; b < B; ++b) {
      const float* qb = qs + (long)b * D + lane * E;
      float acc = 0.f;
#pragma unroll
      for (int e = 0; e < E; ++e) acc += rv[e] * qb[e];
      acc = wave_sum(acc);
      if (lane == 0) scores[(long)b * N + row] = acc;
    }
  }
}

template <typename T>
hipError_t launch_cosine(const void* mat, const void* q, float* scores, long N,
                         int D, int B, hipStream_t stream) {
  if (D % WAVE != 0) return hipErrorInvalidValue;
  size_t lds = (size_t)B * D * sizeof(float);
  if (lds > 160 * 1024) return hipErrorInvalidValue;
  long blocks = (N + 3) / 4;
  if (blocks > 4096) blocks = 4096;
  dim3 g((unsigned)blocks), blk(256);
  switch (D / WAVE) {
#define CASE_E(EV)                                                         \
  case EV:                                                                 \
    hipLaunchKernelGGL((cosine_scores_kernel<T, EV>), g, blk, lds, stream, \
                       (const T*)mat, (const T*)q, scores, N, B);          \
    break;
    CASE_E(1) CASE_E(2) CASE_E(4) CASE_E(6) CASE_E(8)
    CASE_E(12) CASE_E(16) CASE_E(24) CASE_E(32)
#undef CASE_E
    default:
      return hipErrorInvalidValue;
  }
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

// ---------------------------------------------------------------- BM25
// Grid-stride over the query's postings union.  For posting j of term t:
//   contrib = idf[t] * (tf*(k1+1)/(tf + k1*(1-b+b*dl/avgdl)) + delta)
// atomically accumulated into scores[doc].  T (query terms) is small, so a
// linear scan finds the owning term.
__global__ void bm25_kernel(const long* __restrict__ term_ids,
                            const long* __restrict__ qoff,   // [T+1] prefix
                            const long* __restrict__ starts, // [T] indptr[t]
                            const int* __restrict__ post_doc,
                            const float* __restrict__ post_tf,
                            const float* __restrict__ idf,
                            const float* __restrict__ doc_len,
                            float* __restrict__ scores,
                            int T, long total, float k1, float b,
                            float inv_avgdl, float delta) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int t = 0;
    while (t + 1 < T && i >= qoff[t + 1]) ++t;
    const long p = starts[t] + (i - qoff[t]);
    const int doc = post_doc[p];
    const float tf = post_tf[p];
    const float denom = tf + k1 * (1.f - b + b * doc_len[doc] * inv_avgdl);
    const float contrib =
        idf[term_ids[t]] * (tf * (k1 + 1.f) / denom + delta);
    atomicAdd(&scores[doc], contrib);
  }
}


// ------------------------------------------------------------- fusion (K4)
// Batched device fusion of per-source top-k candidate lists
// (reference hybrid.py:204-259 semantics; host fusion.py is the oracle).
// One block per query; union <= FUSE_MAX entries; ids are int64 (-1 = pad).
// method: 0=rrf, 1=weighted_rrf, 2=comb_sum.
#define FUSE_MAX 128
__global__ void fuse_topk_kernel(
    const long* __restrict__ d_ids, const float* __restrict__ d_scores,
    const long* __restrict__ s_ids, const float* __restrict__ s_scores,
    long* __restrict__ out_ids, float* __restrict__ out_scores,
    int Kd, int Ks, int top_k, int method, float rrf_k, float dw, float sw) {
  __shared__ long ids[FUSE_MAX];
  __shared__ float score[FUSE_MAX];
  __shared__ unsigned char owner[FUSE_MAX];
  __shared__ float red[2];
  const int q = blockIdx.x;
  const int n = Kd + Ks;
  const int t = threadIdx.x;

  // load candidates in insertion order: dense ranks then sparse ranks
  for (int i = t; i < n; i += blockDim.x) {
    ids[i] = (i < Kd) ? d_ids[(long)q * Kd + i] : s_ids[(long)q * Ks + (i - Kd)];
    score[i] = 0.f;
    owner[i] = 0;
  }
  __syncthreads();
  // owner = first slot with this id (stable insertion order)
  for (int i = t; i < n; i += blockDim.x) {
    if (ids[i] < 0) continue;
    int first = i;
    for (int j = 0; j < i; ++j)
      if (ids[j] == ids[i]) { first = j; break; }
    owner[i] = (first == i);
  }
  __syncthreads();

  // per-source normalization constants (comb_sum only)
  float dmin = 0.f, dscale = 0.f, smin = 0.f, sscale = 0.f;
  if (method == 2) {
    if (t == 0) {
      float lo = INFINITY, hi = -INFINITY;
      for (int i = 0; i < Kd; ++i) {
        const long id = d_ids[(long)q * Kd + i];
        if (id < 0) continue;
        const float v = d_scores[(long)q * Kd + i];
        lo = fminf(lo, v); hi = fmaxf(hi, v);
      }
      red[0] = lo; red[1] = hi;
    }
    __syncthreads();
    dmin = red[0]; dscale = red[1] - red[0];
    __syncthreads();
    if (t == 0) {
      float lo = INFINITY, hi = -INFINITY;
      for (int i = 0; i < Ks; ++i) {
        const long id = s_ids[(long)q * Ks + i];
        if (id < 0) continue;
        const float v = s_scores[(long)q * Ks + i];
        lo = fminf(lo, v); hi = fmaxf(hi, v);
      }
      red[0] = lo; red[1] = hi;
    }
    __syncthreads();
    smin = red[0]; sscale = red[1] - red[0];
  }

  // accumulate each entry's contribution into its owner slot
  for (int i = t; i < n; i += blockDim.x) {
    if (ids[i] < 0) continue;
    const bool is_dense = i < Kd;
    const int rank = is_dense ? i : (i - Kd);
    float c;
    if (method == 2) {
      const float raw = is_dense ? d_scores[(long)q * Kd + rank]
                                 : s_scores[(long)q * Ks + rank];
      const float mn = is_dense ? dmin : smin;
      const float sc = is_dense ? dscale : sscale;
      const float norm = (sc > 0.f) ? (raw - mn) / sc : 1.f;
      c = (is_dense ? dw : sw) * norm;
    } else {
      const float w = (method == 0) ? 1.f : (is_dense ? dw : sw);
      c = w / (rrf_k + (float)rank);
    }
    // find owner slot
    int o = i;
    for (int j = 0; j < i; ++j)
      if (ids[j] == ids[i]) { o = j; break; }
    atomicAdd(&score[o], c);
  }
  __syncthreads();

  // top-k selection: thread 0 scans (n <= 128, k <= 20 — trivial)
  if (t == 0) {
    for (int k = 0; k < top_k; ++k) {
      float best = -INFINITY;
      int bi = -1;
      for (int i = 0; i < n; ++i)
        if (owner[i] && score[i] > best) { best = score[i]; bi = i; }
      if (bi < 0) {
        out_ids[(long)q * top_k + k] = -1;
        out_scores[(long)q * top_k + k] = 0.f;
      } else {
        out_ids[(long)q * top_k + k] = ids[bi];
        out_scores[(long)q * top_k + k] = best;
        owner[bi] = 0;
      }
    }
  }
}

extern "C" {

hipError_t sentio_cosine_scores_f16(const void* mat, const void* q,
                                    float* scores, long N, int D, int B,
                                    hipStream_t stream) {
  return launch_cosine<_Float16>(mat, q, scores, N, D, B, stream);
}

hipError_t sentio_cosine_scores_bf16(const void* mat, const void* q,
                                     float* scores, long N, int D, int B,
                                     hipStream_t stream) {
  return launch_cosine<__bf16>(mat, q, scores, N, D, B, stream);
}

hipError_t sentio_bm25(const long* term_ids, const long* qoff,
                       const long* starts, const int* post_doc,
                       const float* post_tf, const float* idf,
                       const float* doc_len, float* scores, int T, long total,
                       float k1, float b, float avgdl, float delta,
                       hipStream_t stream) {
  if (total == 0) return hipSuccess;
  long blocks = (total + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  hipLaunchKernelGGL(bm25_kernel, dim3((unsigned)blocks), dim3(256), 0, stream,
                     term_ids, qoff, starts, post_doc, post_tf, idf, doc_len,
                     scores, T, total, k1, b, 1.f / avgdl, delta);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sentio_fuse_topk(const long* d_ids, const float* d_scores,
                            const long* s_ids, const float* s_scores,
                            long* out_ids, float* out_scores, int B, int Kd,
                            int Ks, int top_k, int method, float rrf_k,
                            float dw, float sw, hipStream_t stream) {
  if (Kd + Ks > FUSE_MAX || top_k > FUSE_MAX) return hipErrorInvalidValue;
  hipLaunchKernelGGL(fuse_topk_kernel, dim3(B), dim3(128), 0, stream, d_ids,
                     d_scores, s_ids, s_scores, out_ids, out_scores, Kd, Ks,
                     top_k, method, rrf_k, dw, sw);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

}  // extern "C"
