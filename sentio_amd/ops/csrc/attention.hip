// Attention kernels for gfx950 (CDNA4):
//  * flash_attn  — prefill: flash-style online-softmax attention on
//    mfma_f32_16x16x32_bf16, never materializing the S×S score matrix.
//    One wave per (batch, head, 16-row Q tile); K staged XOR-swizzled in LDS
//    (row-major D-stride tiles are an up-to-16-way bank conflict — guide §6
//    G4), V staged transposed so the PV B-fragment is a contiguous
//    ds_read_b128, P round-trips through padded LDS for the C→A relayout.
//  * decode_attn — single-token decode: one block per (batch, head),
//    chunked online softmax over the contiguous KV cache; score phase is
//    thread-per-key, PV phase is thread-per-dim (coalesced V reads).
//
// Replaces (K6 prefill/decode in SURVEY §2.3) the reference's remote
// /chat/completions calls (reference src/core/llm/providers/openai.py:117).
#include "common.h"

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

// MFMA fragment maps for mfma_f32_16x16x32_bf16 (guide §3):
//   A[16m x 32k]: lane l -> row = l&15,  k = (l>>4)*8 + j   (j = 0..7)
//   B[32k x 16n]: lane l -> col = l&15,  k = (l>>4)*8 + j
//   C/D[16m x 16n]: lane l, reg r -> col = l&15, row = (l>>4)*4 + r

#define QBLK 16
#define KVBLK 32
#define MAXD 128

// LDS byte-offset helpers
// K tile [KVBLK][D] bf16, XOR-swizzled: byte ^= (row&7)<<4
__device__ __forceinline__ int k_lds_off(int row, int d_byte, int Dbytes) {
  return (row * Dbytes + d_byte) ^ ((row & 7) << 4);
}
// V^T tile [D][KVBLK] bf16 with 8-byte row pad: row stride = 64+8 = 72 B
#define VT_STRIDE 72
__device__ __forceinline__ int vt_lds_off(int d, int key_byte) {
  return d * VT_STRIDE + key_byte;
}
// P tile [QBLK][KVBLK] bf16 with 16-byte row pad: stride = 64+16 = 80 B
#define P_STRIDE 80

__launch_bounds__(64)
__global__ void flash_attn_kernel(
    const bf16* __restrict__ q,    // [B, S, H, D]
    const bf16* __restrict__ k,    // [B, S, Hkv, D]
    const bf16* __restrict__ v,    // [B, S, Hkv, D]
    bf16* __restrict__ out,        // [B, S, H, D]
    const int* __restrict__ kv_lens,  // [B]
    int B, int S, int H, int Hkv, int D, float scale, int causal) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // carve: K tile | V^T tile | P tile
  char* k_lds = smem;                                   // KVBLK * D * 2
  char* vt_lds = k_lds + KVBLK * D * 2;                 // D * VT_STRIDE
  char* p_lds = vt_lds + D * VT_STRIDE;                 // QBLK * P_STRIDE

  const int lane = threadIdx.x;
  const int qt = blockIdx.x;            // q tile index
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (H / Hkv);
  const int q0 = qt * QBLK;
  const int kvlen = min(kv_lens[b], S);
  if (q0 >= S) return;

  const int DC = D / 32;                // feature chunks per mfma K-dim
  const int NB = D / 16;                // output column blocks

  // ---- load Q fragments: a_q[dc] = Q[q0 + (l&15)][dc*32 + (l>>4)*8 + j]
  const int arow = lane & 15;
  const int kofs = (lane >> 4) * 8;
  bf16x8_t a_q[4];
  {
    const int qrow = q0 + arow;
    const bf16* qp = q + (((long)b * S + qrow) * H + h) * D + kofs;
#pragma unroll
    for (int dc = 0; dc < 4; ++dc) {
      if (dc < DC) {
        if (qrow < S) {
          a_q[dc] = *reinterpret_cast<const bf16x8_t*>(qp + dc * 32);
        } else {
          bf16x8_t z = {};
          a_q[dc] = z;
        }
      }
    }
  }

  // ---- accumulators
  f32x4_t o_acc[MAXD / 16];             // O in C-frag layout per 16-col block
#pragma unroll
  for (int nb = 0; nb < MAXD / 16; ++nb) o_acc[nb] = f32x4_t{};
  // per-lane row state: the C layout puts row = (l>>4)*4 + r; softmax rows
  // are shared by the 16 lanes of each group -> track per (group,reg)
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -INFINITY; l_run[r] = 0.f; }

  const int kv_hi = causal ? min(kvlen, q0 + QBLK) : kvlen;
  const int n_kv_tiles = (kv_hi + KVBLK - 1) / KVBLK;

  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    const int kv0 = kt * KVBLK;
    // ---- stage K tile (swizzled) and V^T tile into LDS (one wave)
    // K: rows kv0..kv0+31; each lane stages rows lane/2 (2 lanes per row,
    // each lane covers D/2 bytes when D=128 → use generic loop
    {
      const int elems = KVBLK * D;      // bf16 elements in tile
      for (int i = lane * 8; i < elems; i += WAVE * 8) {
        const int row = i / D;
        const int d = i % D;
        const int key = kv0 + row;
        bf16x8_t val;
        if (key < kvlen) {
          val = *reinterpret_cast<const bf16x8_t*>(
              k + (((long)b * S + key) * Hkv + hkv) * D + d);
        } else {
          bf16x8_t z = {};
          val = z;
        }
        // swizzled K write: 16B aligned chunks keep XOR validity ((d*2)%16==0)
        *reinterpret_cast<bf16x8_t*>(k_lds + k_lds_off(row, d * 2, D * 2)) = val;
        // V^T write: transpose — scalar stores
        bf16x8_t vv;
        if (key < kvlen) {
          vv = *reinterpret_cast<const bf16x8_t*>(
              v + (((long)b * S + key) * Hkv + hkv) * D + d);
        } else {
          bf16x8_t z = {};
          vv = z;
        }
#pragma unroll
        for (int j = 0; j < 8; ++j)
          *reinterpret_cast<__bf16*>(vt_lds + vt_lds_off(d + j, row * 2)) = vv[j];
      }
    }
    __syncthreads();  // single wave: compiles to s_waitcnt; keeps LDS ordered

    // ---- S = scale * Q K^T for the two 16-key halves
    f32x4_t s_acc[2];
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      s_acc[half] = f32x4_t{};
#pragma unroll
      for (int dc = 0; dc < 4; ++dc) {
        if (dc < DC) {
          // B frag: K_lds[half*16 + (l&15)][dc*32 + kofs + j]
          const int krow = half * 16 + (lane & 15);
          bf16x8_t b_frag = *reinterpret_cast<const bf16x8_t*>(
              k_lds + k_lds_off(krow, (dc * 32 + kofs) * 2, D * 2));
          s_acc[half] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_q[dc], b_frag, s_acc[half], 0, 0, 0);
        }
      }
    }

    // ---- online softmax on the 16x32 score tile
    // lane holds: col = l&15 (+16*half), rows = (l>>4)*4 + r
    float p_val[2][4];
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow_local = (lane >> 4) * 4 + r;
      const int qrow = q0 + qrow_local;
      float s0 = s_acc[0][r] * scale;
      float s1 = s_acc[1][r] * scale;
      const int key0 = kv0 + (lane & 15);
      const int key1 = key0 + 16;
      bool ok0 = key0 < kvlen && (!causal || key0 <= qrow);
      bool ok1 = key1 < kvlen && (!causal || key1 <= qrow);
      s0 = ok0 ? s0 : -INFINITY;
      s1 = ok1 ? s1 : -INFINITY;
      // row max across the 16 lanes of the group (both halves)
      float rmax = group16_max(fmaxf(s0, s1));
      float m_new = fmaxf(m_run[r], rmax);
      // guard: fully-masked row keeps m=-inf; exp(-inf - -inf) handled below
      float a = (m_run[r] == -INFINITY) ? 0.f
                : __expf(m_run[r] - m_new);
      if (m_new == -INFINITY) a = 1.f;  // nothing seen yet at all
      alpha[r] = a;
      float p0 = (s0 == -INFINITY) ? 0.f : __expf(s0 - m_new);
      float p1 = (s1 == -INFINITY) ? 0.f : __expf(s1 - m_new);
      float rsum = group16_sum(p0 + p1);
      l_run[r] = l_run[r] * a + rsum;
      m_run[r] = m_new;
      p_val[0][r] = p0;
      p_val[1][r] = p1;
    }

    // ---- write P to LDS in C layout, reread as A fragments
    __syncthreads();
#pragma unroll
    for (int half = 0; half < 2; ++half) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow_local = (lane >> 4) * 4 + r;
        const int key_local = half * 16 + (lane & 15);
        *reinterpret_cast<__bf16*>(
            p_lds + qrow_local * P_STRIDE + key_local * 2) =
            (__bf16)p_val[half][r];
      }
    }
    __syncthreads();
    // A frag for PV: P[row = l&15][k = key = kofs + j] — one frag covers all
    // 32 keys of the tile (k = (l>>4)*8 + j spans 0..31)
    const bf16x8_t a_p = *reinterpret_cast<const bf16x8_t*>(
        p_lds + (lane & 15) * P_STRIDE + kofs * 2);

    // ---- O = alpha*O + P V   (one mfma per 16-col block of V)
#pragma unroll
    for (int nb = 0; nb < MAXD / 16; ++nb) {
      if (nb < NB) {
        // rescale accumulator rows by alpha[r]
#pragma unroll
        for (int r = 0; r < 4; ++r) o_acc[nb][r] *= alpha[r];
        // B frag: V^T[d = nb*16 + (l&15)][key = kofs + j]
        bf16x8_t b_v;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          b_v[j] = *reinterpret_cast<const __bf16*>(
              vt_lds + vt_lds_off(nb * 16 + (lane & 15), (kofs + j) * 2));
        o_acc[nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_p, b_v, o_acc[nb], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: divide by l, store
#pragma unroll
  for (int nb = 0; nb < MAXD / 16; ++nb) {
    if (nb < NB) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow_local = (lane >> 4) * 4 + r;
        const int qrow = q0 + qrow_local;
        if (qrow < S) {
          const float denom = l_run[r] > 0.f ? l_run[r] : 1.f;
          const int d = nb * 16 + (lane & 15);
          out[(((long)b * S + qrow) * H + h) * D + d] =
              f2bf(o_acc[nb][r] / denom);
        }
      }
    }
  }
}

// ------------------------------------------------------------ decode attn
// q [B, H, D], k/v cache [B, Hkv, Smax, D], seq_lens [B] -> out [B, H, D].
// Block (256 threads) per (b, h).  Chunked online softmax:
//   phase A: thread-per-key dot products (vectorized K row reads)
//   phase B: thread-per-dim PV accumulation (coalesced V reads)
#define DEC_CHUNK 256

__global__ void decode_attn_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ kc,
    const bf16* __restrict__ vc, bf16* __restrict__ out,
    const int* __restrict__ seq_lens,
    int H, int Hkv, int Smax, int D, float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* p_sh = reinterpret_cast<float*>(smem);            // [DEC_CHUNK]
  float* q_sh = p_sh + DEC_CHUNK;                          // [D]
  float* red = q_sh + D;                                   // [32] scratch

  const int h = blockIdx.x;
  const int b = blockIdx.y;
  const int hkv = h / (H / Hkv);
  const int slen = seq_lens[b];
  const bf16* qb = q + ((long)b * H + h) * D;
  const bf16* kb = kc + ((long)b * Hkv + hkv) * Smax * (long)D;
  const bf16* vb = vc + ((long)b * Hkv + hkv) * Smax * (long)D;

  for (int d = threadIdx.x; d < D; d += blockDim.x) q_sh[d] = bf2f(qb[d]);
  __syncthreads();

  float m_run = -INFINITY, l_run = 0.f;
  // per-thread O accumulators over dims (D <= 256 with 256 threads)
  float o0 = 0.f;
  const int myd = threadIdx.x;          // dim owned in phase B (if < D)

  for (int s0 = 0; s0 < slen; s0 += DEC_CHUNK) {
    const int chunk = min(DEC_CHUNK, slen - s0);
    // phase A: dot for own key
    float sc = -INFINITY;
    const int s = s0 + threadIdx.x;
    if (threadIdx.x < chunk) {
      const bf16* krow = kb + (long)s * D;
      float acc = 0.f;
      for (int d = 0; d < D; d += 8) {
        bf16x8 kv8 = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const short*>(krow) + d);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc += bits2f(kv8[j]) * q_sh[d + j];
      }
      sc = acc * scale;
    }
    // chunk max
    float cmax = block_max(sc, red);
    float m_new = fmaxf(m_run, cmax);
    float alpha = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
    if (m_new == -INFINITY) alpha = 1.f;
    float p = (threadIdx.x < chunk && sc != -INFINITY) ? __expf(sc - m_new) : 0.f;
    p_sh[threadIdx.x] = p;
    float csum = block_sum(p, red);
    l_run = l_run * alpha + csum;
    m_run = m_new;
    __syncthreads();
    // phase B: PV for own dim
    if (myd < D) {
      float acc = 0.f;
      for (int j = 0; j < chunk; ++j)
        acc += p_sh[j] * bf2f(vb[(long)(s0 + j) * D + myd]);
      o0 = o0 * alpha + acc;
    }
    __syncthreads();
  }

  if (myd < D) {
    const float denom = l_run > 0.f ? l_run : 1.f;
    out[((long)b * H + h) * D + myd] = f2bf(o0 / denom);
  }
}

extern "C" {

hipError_t sentio_flash_attn(const void* q, const void* k, const void* v,
                             void* out, const int* kv_lens, int B, int S,
                             int H, int Hkv, int D, float scale, int causal,
                             hipStream_t stream) {
  if (D % 32 != 0 || D > MAXD) return hipErrorInvalidValue;
  size_t lds = (size_t)KVBLK * D * 2 + (size_t)D * VT_STRIDE + QBLK * P_STRIDE;
  dim3 grid((S + QBLK - 1) / QBLK, H, B);
  hipLaunchKernelGGL(flash_attn_kernel, grid, dim3(64), lds, stream,
                     (const bf16*)q, (const bf16*)k, (const bf16*)v,
                     (bf16*)out, kv_lens, B, S, H, Hkv, D, scale, causal);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sentio_decode_attn(const void* q, const void* kc, const void* vc,
                              void* out, const int* seq_lens, int B, int H,
                              int Hkv, int Smax, int D, float scale,
                              hipStream_t stream) {
  if (D > 256) return hipErrorInvalidValue;
  size_t lds = (DEC_CHUNK + D + 32) * sizeof(float);
  dim3 grid(H, B);
  hipLaunchKernelGGL(decode_attn_kernel, grid, dim3(DEC_CHUNK), lds, stream,
                     (const bf16*)q, (const bf16*)kc, (const bf16*)vc,
                     (bf16*)out, seq_lens, H, Hkv, Smax, D, scale);
  HIP_CHECK_LAUNCH();
  return hipSuccess;
}

}  // extern "C"
