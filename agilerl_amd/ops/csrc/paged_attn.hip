// Paged-attention decode kernels (CDNA4, bf16 pools, fp32 math).
//
// Single-token decode attention reading K/V straight from the paged pool
// through per-sequence page tables — no gather, no contiguous copy.
// Decode attention is bandwidth-bound: the split kernel is a
// flash-decoding structure sized for HBM3E —
//
//   grid (B, Hkv, splits), block 256 (4 waves).  Within a block, lanes are
//   tiled into GROUPS of lpt = D/8 lanes; each group owns one token per
//   iteration and each lane issues ONE 16-byte load (8 bf16) for its slice
//   of K and of V, so a block streams 16 tokens x 2 x D x 2B per iteration
//   in fully-coalesced 16B transactions.  GQA query heads ride along in
//   registers.  Per-group online softmax; xor-shuffle combine across
//   groups in a wave; LDS combine across waves; per-split partials
//   (m, l, acc) land in a workspace and a second tiny kernel reduces
//   splits.  The page table row is staged in LDS.
//
// Shapes: D in {16, 32, 64, 128} (8 * pow2 lanes-per-token), GQA <= 8,
// page_size pow2.  A scalar fallback kernel covers anything else.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define MAX_GQA 16
#define FD_BLOCK 256
#define FD_WAVES (FD_BLOCK / WAVE)
#define FD_MAX_GQA 8
#define FD_MAX_D 128
#define FD_MAX_SPLITS 32
#define FD_LDS_TABLE 2048

typedef __hip_bfloat16 bf16;

__device__ inline float pa_bf2f(bf16 v) { return __bfloat162float(v); }

// 8 bf16 -> 8 floats via one 16-byte load
__device__ inline void load_bf16x8(const bf16* __restrict__ p, float* f) {
  ushort v[8];
  *reinterpret_cast<int4*>(v) = *reinterpret_cast<const int4*>(p);
  #pragma unroll
  for (int i = 0; i < 8; ++i) {
    unsigned int u = ((unsigned int)v[i]) << 16;
    f[i] = __uint_as_float(u);
  }
}

__device__ inline float wave_sum(float v) {
  #pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

// ---------------------------------------------------------------------------
// Split kernel: one block per (seq, kv-head, split).
// Workspace: ws_acc (B, Hq, splits, D) fp32; ws_ml (B, Hq, splits, 2).
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(FD_BLOCK)
paged_attn_split_kernel(
    const bf16* __restrict__ q,
    const bf16* __restrict__ k_pool, const bf16* __restrict__ v_pool,
    const int* __restrict__ table, const int* __restrict__ lengths,
    float* __restrict__ ws_acc, float* __restrict__ ws_ml,
    int B, int Hq, int Hkv, int D, int log2S, int max_pages, int splits,
    float scale) {
  int b = blockIdx.x;
  int hk = blockIdx.y;
  int split = blockIdx.z;
  int gqa = Hq / Hkv;
  int len = lengths[b];
  int S = 1 << log2S;

  __shared__ int lds_table[FD_LDS_TABLE];
  __shared__ float lds_m[FD_WAVES][FD_MAX_GQA];
  __shared__ float lds_l[FD_WAVES][FD_MAX_GQA];
  __shared__ float lds_acc[FD_WAVES][FD_MAX_GQA][FD_MAX_D];

  int tid = threadIdx.x;
  int n_table = min(max_pages, FD_LDS_TABLE);
  for (int i = tid; i < n_table; i += FD_BLOCK)
    lds_table[i] = table[(long)b * max_pages + i];
  __syncthreads();

  // split token range (balanced per sequence)
  int chunk = (len + splits - 1) / splits;
  int s0 = split * chunk;
  int s1 = min(s0 + chunk, len);

  int lpt = D / 8;                 // lanes per token
  int lane = tid & (WAVE - 1);
  int wav = tid / WAVE;
  int grp_in_wave = lane / lpt;    // token-group within wave
  int p = lane & (lpt - 1);        // position in group -> elem slice [8p, 8p+8)
  int groups_per_wave = WAVE / lpt;
  int G = FD_WAVES * groups_per_wave;           // tokens per block-iteration
  int grp = wav * groups_per_wave + grp_in_wave;

  // q slices for every GQA head this kv-head serves
  float qv[FD_MAX_GQA][8];
  for (int g = 0; g < gqa; ++g)
    load_bf16x8(q + ((long)b * Hq + hk * gqa + g) * D + p * 8, qv[g]);

  float m[FD_MAX_GQA], l[FD_MAX_GQA], acc[FD_MAX_GQA][8];
  for (int g = 0; g < gqa; ++g) {
    m[g] = -1e30f; l[g] = 0.f;
    #pragma unroll
    for (int e = 0; e < 8; ++e) acc[g][e] = 0.f;
  }

  // software pipeline: issue the NEXT token's K/V loads before the current
  // token's shuffle-reduce/exp chain, hiding HBM latency behind ALU work
  auto load_kv = [&](int t, float* kf, float* vf) {
    int page = (t >> log2S) < n_table ? lds_table[t >> log2S]
                                      : table[(long)b * max_pages + (t >> log2S)];
    long base = (((long)page * S + (t & (S - 1))) * Hkv + hk) * D + p * 8;
    load_bf16x8(k_pool + base, kf);
    load_bf16x8(v_pool + base, vf);
  };
  float kf[8], vf[8];
  int t0 = s0 + grp;
  if (t0 < s1) load_kv(t0, kf, vf);
  for (int t = t0; t < s1; t += G) {
    float kf2[8], vf2[8];
    if (t + G < s1) load_kv(t + G, kf2, vf2);
    for (int g = 0; g < gqa; ++g) {
      float part = 0.f;
      #pragma unroll
      for (int e = 0; e < 8; ++e) part += qv[g][e] * kf[e];
      // reduce across the lpt lanes of this group
      for (int off = lpt >> 1; off > 0; off >>= 1) part += __shfl_xor(part, off);
      float s = part * scale;
      float m_new = fmaxf(m[g], s);
      float corr = __expf(m[g] - m_new);
      float pr = __expf(s - m_new);
      l[g] = l[g] * corr + pr;
      #pragma unroll
      for (int e = 0; e < 8; ++e) acc[g][e] = acc[g][e] * corr + pr * vf[e];
      m[g] = m_new;
    }
    #pragma unroll
    for (int e = 0; e < 8; ++e) { kf[e] = kf2[e]; vf[e] = vf2[e]; }
  }

  // combine groups within the wave: xor offsets lpt..WAVE/2 keep the
  // elem-slice position p, pairing partials of different token subsets
  for (int off = lpt; off < WAVE; off <<= 1) {
    for (int g = 0; g < gqa; ++g) {
      float m2 = __shfl_xor(m[g], off);
      float l2 = __shfl_xor(l[g], off);
      float m12 = fmaxf(m[g], m2);
      float c1 = __expf(m[g] - m12);
      float c2 = __expf(m2 - m12);
      #pragma unroll
      for (int e = 0; e < 8; ++e) {
        float a2 = __shfl_xor(acc[g][e], off);
        acc[g][e] = acc[g][e] * c1 + a2 * c2;
      }
      l[g] = l[g] * c1 + l2 * c2;
      m[g] = m12;
    }
  }

  // wave partial -> LDS (lanes 0..lpt-1 hold the full wave result)
  if (lane < lpt) {
    for (int g = 0; g < gqa; ++g) {
      if (lane == 0) { lds_m[wav][g] = m[g]; lds_l[wav][g] = l[g]; }
      #pragma unroll
      for (int e = 0; e < 8; ++e) lds_acc[wav][g][p * 8 + e] = acc[g][e];
    }
  }
  __syncthreads();

  // wave 0 combines the FD_WAVES partials and writes the split result
  if (wav == 0 && lane < lpt) {
    for (int g = 0; g < gqa; ++g) {
      float mm = lds_m[0][g];
      for (int w = 1; w < FD_WAVES; ++w) mm = fmaxf(mm, lds_m[w][g]);
      float ll = 0.f;
      float oacc[8];
      #pragma unroll
      for (int e = 0; e < 8; ++e) oacc[e] = 0.f;
      for (int w = 0; w < FD_WAVES; ++w) {
        float c = __expf(lds_m[w][g] - mm);
        ll += lds_l[w][g] * c;
        #pragma unroll
        for (int e = 0; e < 8; ++e) oacc[e] += lds_acc[w][g][p * 8 + e] * c;
      }
      long h = (long)b * Hq + hk * gqa + g;
      long wa = (h * splits + split) * D + p * 8;
      #pragma unroll
      for (int e = 0; e < 8; ++e) ws_acc[wa + e] = oacc[e];
      if (lane == 0) {
        ws_ml[(h * splits + split) * 2 + 0] = mm;
        ws_ml[(h * splits + split) * 2 + 1] = ll;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Reduce kernel: combine split partials.  grid (B*Hq), block D.
// ---------------------------------------------------------------------------
__global__ void paged_attn_reduce_kernel(
    const float* __restrict__ ws_acc, const float* __restrict__ ws_ml,
    float* __restrict__ out, int splits, int D) {
  long h = blockIdx.x;
  int d = threadIdx.x;
  if (d >= D) return;
  float m_star = -1e30f;
  for (int s = 0; s < splits; ++s)
    m_star = fmaxf(m_star, ws_ml[(h * splits + s) * 2 + 0]);
  float denom = 0.f, num = 0.f;
  for (int s = 0; s < splits; ++s) {
    float c = __expf(ws_ml[(h * splits + s) * 2 + 0] - m_star);
    denom += ws_ml[(h * splits + s) * 2 + 1] * c;
    num += ws_acc[(h * splits + s) * D + d] * c;
  }
  out[h * D + d] = num / denom;
}

// ---------------------------------------------------------------------------
// Scalar fallback (original wave-reduction kernel) for shapes outside the
// flash-decoding constraints.
// ---------------------------------------------------------------------------
__global__ void paged_attn_basic_kernel(
    const bf16* __restrict__ q,
    const bf16* __restrict__ k_pool, const bf16* __restrict__ v_pool,
    const int* __restrict__ table, const int* __restrict__ lengths,
    float* __restrict__ out,
    int B, int Hq, int Hkv, int D, int S, int max_pages, float scale) {
  int b = blockIdx.x;
  int hk = blockIdx.y;
  int gqa = Hq / Hkv;
  int lane = threadIdx.x;
  int len = lengths[b];
  if (b >= B || len <= 0) return;

  float qv[MAX_GQA][4];
  int elems = 0;
  for (int j = lane; j < D; j += WAVE) {
    for (int g = 0; g < gqa; ++g)
      qv[g][elems] = pa_bf2f(q[((long)b * Hq + hk * gqa + g) * D + j]);
    ++elems;
  }
  float m[MAX_GQA], l[MAX_GQA], acc[MAX_GQA][4];
  for (int g = 0; g < gqa; ++g) {
    m[g] = -1e30f; l[g] = 0.f;
    for (int e = 0; e < 4; ++e) acc[g][e] = 0.f;
  }
  for (int t = 0; t < len; ++t) {
    int page = table[(long)b * max_pages + t / S];
    long base = (((long)page * S + (t % S)) * Hkv + hk) * D;
    float kv[4], vv[4];
    int e = 0;
    for (int j = lane; j < D; j += WAVE) {
      kv[e] = pa_bf2f(k_pool[base + j]);
      vv[e] = pa_bf2f(v_pool[base + j]);
      ++e;
    }
    for (int g = 0; g < gqa; ++g) {
      float part = 0.f;
      for (int i = 0; i < elems; ++i) part += qv[g][i] * kv[i];
      float s = wave_sum(part) * scale;
      float m_new = fmaxf(m[g], s);
      float corr = __expf(m[g] - m_new);
      float p = __expf(s - m_new);
      l[g] = l[g] * corr + p;
      for (int i = 0; i < elems; ++i) acc[g][i] = acc[g][i] * corr + p * vv[i];
      m[g] = m_new;
    }
  }
  for (int g = 0; g < gqa; ++g) {
    float inv = 1.f / l[g];
    int e = 0;
    for (int j = lane; j < D; j += WAVE) {
      out[((long)b * Hq + hk * gqa + g) * D + j] = acc[g][e] * inv;
      ++e;
    }
  }
}

static inline bool is_pow2(int x) { return x > 0 && (x & (x - 1)) == 0; }

torch::Tensor paged_attn_decode(
    torch::Tensor q, torch::Tensor k_pool, torch::Tensor v_pool,
    torch::Tensor table, torch::Tensor lengths, double scale,
    long max_len_hint) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16,
              "paged_attn: bf16 cuda q required");
  TORCH_CHECK(q.dim() == 3 && k_pool.dim() == 4, "paged_attn: bad shapes");
  auto qc = q.contiguous();
  auto kc = k_pool.contiguous();
  auto vc = v_pool.contiguous();
  auto tc = table.to(torch::kInt).contiguous();
  auto lc = lengths.to(torch::kInt).contiguous();
  int B = qc.size(0), Hq = qc.size(1), D = qc.size(2);
  int S = kc.size(1), Hkv = kc.size(2);
  int max_pages = tc.size(1);
  int gqa = Hq / Hkv;
  TORCH_CHECK(Hq % Hkv == 0 && gqa <= MAX_GQA, "paged_attn: bad GQA ratio");
  TORCH_CHECK(D <= 4 * WAVE, "paged_attn: head_dim must be <= 256");
  auto out = torch::empty({B, Hq, D}, qc.options().dtype(torch::kFloat));
  hipStream_t stream = at::hip::getCurrentHIPStream();

  bool fd_ok = is_pow2(S) && is_pow2(D / 8) && D % 8 == 0 && D <= FD_MAX_D
               && gqa <= FD_MAX_GQA;
  if (fd_ok) {
    int log2S = 0;
    while ((1 << log2S) < S) ++log2S;
    // block-count target: enough (b, hk, split) blocks to cover the 1024
    // wave slots with latency-hiding headroom, bounded by ~64 tokens per
    // split so short sequences don't shred into empty splits.  Tunable for
    // sweeps via AGILERL_PA_SPLIT_TARGET.
    static long target = [] {
      const char* env = std::getenv("AGILERL_PA_SPLIT_TARGET");
      return env ? std::atol(env) : 1024L;
    }();
    long bh = (long)B * Hkv;
    int splits = (int)std::min<long>(FD_MAX_SPLITS, std::max<long>(1, target / bh));
    // bound by ~64 tokens per split using the caller's length hint when
    // given (lengths live on device; max_pages wildly overestimates short
    // decodes — measured 78 us/call from 16 one-iteration splits at
    // len<=256 before the hint existed)
    long len_bound = max_len_hint > 0 ? max_len_hint : (long)max_pages * S;
    splits = (int)std::min<long>(splits, std::max<long>(1, (len_bound + 63) / 64));
    auto ws_acc = torch::empty({(long)B * Hq * splits * D},
                               qc.options().dtype(torch::kFloat));
    auto ws_ml = torch::empty({(long)B * Hq * splits * 2},
                              qc.options().dtype(torch::kFloat));
    hipLaunchKernelGGL(paged_attn_split_kernel,
        dim3(B, Hkv, splits), dim3(FD_BLOCK), 0, stream,
        reinterpret_cast<const bf16*>(qc.data_ptr()),
        reinterpret_cast<const bf16*>(kc.data_ptr()),
        reinterpret_cast<const bf16*>(vc.data_ptr()),
        tc.data_ptr<int>(), lc.data_ptr<int>(),
        ws_acc.data_ptr<float>(), ws_ml.data_ptr<float>(),
        B, Hq, Hkv, D, log2S, max_pages, splits, (float)scale);
    hipLaunchKernelGGL(paged_attn_reduce_kernel,
        dim3(B * Hq), dim3(D), 0, stream,
        ws_acc.data_ptr<float>(), ws_ml.data_ptr<float>(),
        out.data_ptr<float>(), splits, D);
    return out;
  }
  hipLaunchKernelGGL(paged_attn_basic_kernel, dim3(B, Hkv), dim3(WAVE), 0, stream,
      reinterpret_cast<const bf16*>(qc.data_ptr()),
      reinterpret_cast<const bf16*>(kc.data_ptr()),
      reinterpret_cast<const bf16*>(vc.data_ptr()),
      tc.data_ptr<int>(), lc.data_ptr<int>(), out.data_ptr<float>(),
      B, Hq, Hkv, D, S, max_pages, (float)scale);
  return out;
}

void init_paged_attn(pybind11::module_& m) {
  m.def("paged_attn_decode", &paged_attn_decode,
        "single-token paged-attention decode (bf16 pools, fp32 out)",
        pybind11::arg("q"), pybind11::arg("k_pool"), pybind11::arg("v_pool"),
        pybind11::arg("table"), pybind11::arg("lengths"),
        pybind11::arg("scale"), pybind11::arg("max_len_hint") = 0L);
}
