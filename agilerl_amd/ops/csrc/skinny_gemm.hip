// Skinny GEMM for single-token decode: out(M,N) = x(M,K) @ W(N,K)^T.
//
// Decode-time projections are M<=16 rows against multi-MB weight
// matrices — effectively M simultaneous GEMVs, bound by streaming W from
// HBM3E.  Measured on MI355X (demos/bench_kernels.py): hipBLASLt carries
// a ~19 us small-GEMM floor, so the 8 MB GQA k/v projections run at
// ~445 GB/s; this kernel reaches 622 GB/s there and is dispatched for
// N<=2048 only — hipBLASLt keeps larger shapes, where its tiles stream
// at 5.5+ TB/s.
//
//   grid = ceil(N / 4) blocks x 256 threads; the whole block strides K
//   for its 4 W rows (short dependent-load chains even at small N).
//   Per K-chunk every lane issues one 16-byte bf16x8 load per W row
//   (fully-coalesced weight traffic) plus M x-loads that ride L2.
//   fp32 accumulation; xor-shuffle + LDS cross-wave reduce.
//
// Requirements: bf16 x/W, K % 8 == 0, M <= 16. Bias unsupported (Llama
// projections are bias-free); LoRA deltas ride separately.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define SG_MAX_M 16
#define SG_ROWS 4     // W rows per wave
#define SG_WAVES 4    // waves per block

typedef __hip_bfloat16 bf16;

__device__ inline void sg_load8(const bf16* __restrict__ p, float* f) {
  ushort v[8];
  *reinterpret_cast<int4*>(v) = *reinterpret_cast<const int4*>(p);
  #pragma unroll
  for (int i = 0; i < 8; ++i) f[i] = __uint_as_float(((unsigned int)v[i]) << 16);
}

template <int MT>
__global__ void __launch_bounds__(WAVE * SG_WAVES)
skinny_gemm_kernel(
    const bf16* __restrict__ x,   // (MT, K)
    const bf16* __restrict__ w,   // (N, K)
    bf16* __restrict__ out,       // (MT, N)
    long N, int K) {
  // One block owns SG_ROWS consecutive W rows; all 256 lanes stride K
  // together (the whole block is one wide K-reader), so small-N shapes
  // still launch N/4 blocks and the per-wave dependent-load chain is a
  // quarter as long as a wave-per-rows layout.
  int wave = threadIdx.x / WAVE;
  int lane = threadIdx.x % WAVE;
  long n0 = (long)blockIdx.x * SG_ROWS;
  if (n0 >= N) return;
  bool full = (n0 + SG_ROWS) <= N;

  float acc[MT][SG_ROWS];
  #pragma unroll
  for (int m = 0; m < MT; ++m)
    #pragma unroll
    for (int r = 0; r < SG_ROWS; ++r) acc[m][r] = 0.f;

  for (int k0 = (int)threadIdx.x * 8; k0 < K; k0 += WAVE * SG_WAVES * 8) {
    float wf[SG_ROWS][8];
    if (full) {
      #pragma unroll
      for (int r = 0; r < SG_ROWS; ++r) sg_load8(w + (n0 + r) * K + k0, wf[r]);
    } else {
      for (int r = 0; r < SG_ROWS; ++r)
        if (n0 + r < N) sg_load8(w + (n0 + r) * K + k0, wf[r]);
    }
    #pragma unroll
    for (int m = 0; m < MT; ++m) {
      float xf[8];
      sg_load8(x + (long)m * K + k0, xf);
      #pragma unroll
      for (int r = 0; r < SG_ROWS; ++r) {
        float part = 0.f;
        #pragma unroll
        for (int e = 0; e < 8; ++e) part += xf[e] * wf[r][e];
        acc[m][r] += part;
      }
    }
  }

  // wave-level shuffle reduce, then cross-wave LDS combine
  __shared__ float lds[SG_WAVES][MT > 0 ? MT : 1][SG_ROWS];
  #pragma unroll
  for (int m = 0; m < MT; ++m)
    #pragma unroll
    for (int r = 0; r < SG_ROWS; ++r) {
      float v = acc[m][r];
      #pragma unroll
      for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_xor(v, off);
      if (lane == 0) lds[wave][m][r] = v;
    }
  __syncthreads();
  if (wave == 0 && lane == 0) {
    #pragma unroll
    for (int m = 0; m < MT; ++m)
      #pragma unroll
      for (int r = 0; r < SG_ROWS; ++r) {
        if (n0 + r >= N) continue;
        float v = 0.f;
        #pragma unroll
        for (int s = 0; s < SG_WAVES; ++s) v += lds[s][m][r];
        out[(long)m * N + n0 + r] = __float2bfloat16(v);
      }
  }
}

torch::Tensor skinny_gemm(torch::Tensor x, torch::Tensor w) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16,
              "skinny_gemm: bf16 cuda x required");
  TORCH_CHECK(w.scalar_type() == torch::kBFloat16, "skinny_gemm: bf16 W required");
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2, "skinny_gemm: 2-D inputs");
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  int M = xc.size(0), K = xc.size(1);
  long N = wc.size(0);
  TORCH_CHECK(wc.size(1) == K, "skinny_gemm: K mismatch");
  TORCH_CHECK(M >= 1 && M <= SG_MAX_M, "skinny_gemm: M must be in [1, 16]");
  TORCH_CHECK(K % 8 == 0, "skinny_gemm: K must be a multiple of 8");
  auto out = torch::empty({M, N}, xc.options());
  long grid = (N + SG_ROWS - 1) / SG_ROWS;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  const bf16* xp = reinterpret_cast<const bf16*>(xc.data_ptr());
  const bf16* wp = reinterpret_cast<const bf16*>(wc.data_ptr());
  bf16* op = reinterpret_cast<bf16*>(out.data_ptr());
  dim3 g((unsigned)grid), b(WAVE * SG_WAVES);
  switch (M) {
#define SG_CASE(MT) case MT: \
    hipLaunchKernelGGL(skinny_gemm_kernel<MT>, g, b, 0, stream, xp, wp, op, N, K); break;
    SG_CASE(1) SG_CASE(2) SG_CASE(3) SG_CASE(4) SG_CASE(5) SG_CASE(6)
    SG_CASE(7) SG_CASE(8) SG_CASE(9) SG_CASE(10) SG_CASE(11) SG_CASE(12)
    SG_CASE(13) SG_CASE(14) SG_CASE(15) SG_CASE(16)
#undef SG_CASE
    default: TORCH_CHECK(false, "skinny_gemm: unsupported M");
  }
  return out;
}

void init_skinny_gemm(pybind11::module_& m) {
  m.def("skinny_gemm", &skinny_gemm,
        "decode-shape GEMM: (M<=16, K) x (N, K)^T, bf16, weight-stream bound");
}
