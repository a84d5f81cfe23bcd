// Paged-attention decode kernel (CDNA4, bf16 pools, fp32 math).
//
// Round-2 decode-engine milestone 2 (docs/design/round2_perf_plan.md §2):
// single-token decode attention reading K/V straight from the paged pool
// through per-sequence page tables — no gather, no contiguous copy.
// One wave per (sequence, kv-head); GQA query heads ride along in
// registers; online softmax over the token stream.  Decode attention is
// bandwidth-bound and small-shaped, so a wave-reduction structure (no
// MFMA) is the right tool; correctness-first, tuning in round 2.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define MAX_GQA 16

typedef __hip_bfloat16 bf16;

__device__ inline float pa_bf2f(bf16 v) { return __bfloat162float(v); }

__device__ inline float wave_sum(float v) {
  #pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

// q: (B, Hq, D); k_pool/v_pool: (P, S, Hkv, D); table: (B, max_pages);
// lengths: (B,); out: (B, Hq, D) fp32.
__global__ void paged_attn_decode_kernel(
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

  // per-lane slice of the head dim (stride WAVE)
  float qv[MAX_GQA][4];  // up to 4 elems/lane => D <= 256
  int elems = 0;
  for (int j = lane; j < D; j += WAVE) {
    for (int g = 0; g < gqa; ++g) {
      qv[g][elems] = pa_bf2f(q[((long)b * Hq + hk * gqa + g) * D + j]);
    }
    ++elems;
  }

  float m[MAX_GQA], l[MAX_GQA], acc[MAX_GQA][4];
  for (int g = 0; g < gqa; ++g) {
    m[g] = -1e30f;
    l[g] = 0.f;
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
      // online softmax update
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

torch::Tensor paged_attn_decode(
    torch::Tensor q, torch::Tensor k_pool, torch::Tensor v_pool,
    torch::Tensor table, torch::Tensor lengths, double scale) {
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
  TORCH_CHECK(Hq % Hkv == 0 && Hq / Hkv <= MAX_GQA, "paged_attn: bad GQA ratio");
  TORCH_CHECK(D <= 4 * WAVE, "paged_attn: head_dim must be <= 256");
  auto out = torch::empty({B, Hq, D}, qc.options().dtype(torch::kFloat));
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(paged_attn_decode_kernel, dim3(B, Hkv), dim3(WAVE), 0, stream,
      reinterpret_cast<const bf16*>(qc.data_ptr()),
      reinterpret_cast<const bf16*>(kc.data_ptr()),
      reinterpret_cast<const bf16*>(vc.data_ptr()),
      tc.data_ptr<int>(), lc.data_ptr<int>(), out.data_ptr<float>(),
      B, Hq, Hkv, D, S, max_pages, (float)scale);
  return out;
}

void init_paged_attn(pybind11::module_& m) {
  m.def("paged_attn_decode", &paged_attn_decode,
        "single-token paged-attention decode (bf16 pools, fp32 out)");
}
