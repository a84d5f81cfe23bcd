// CDNA4 RMSNorm kernels (bf16 in/out, fp32 accumulation).
//
// Analog of the reference's Liger RMSNorm patches (SURVEY §2.9.13,
// architectures/nemotron_h/liger.py): the HF Llama RMSNorm is an eager
// chain (cast, pow, mean, rsqrt, mul, mul) = multiple HBM round-trips;
// these kernels do one pass each way.  Memory-bound: bf16 loads are
// vectorized 8-wide (guide G13); one block per row, shuffle+LDS reduce.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_bf16.h>

#include <vector>

#define WAVE 64

typedef __hip_bfloat16 bf16;

__device__ inline float bf2f(bf16 v) { return __bfloat162float(v); }
__device__ inline bf16 f2bf(float v) { return __float2bfloat16(v); }

// block-reduce a single float (sum) over up to 1024 threads
__device__ float block_sum(float v, float* red) {
  int tid = threadIdx.x;
  int lane = tid & (WAVE - 1);
  int wid = tid / WAVE;
  #pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_down(v, off);
  if (lane == 0) red[wid] = v;
  __syncthreads();
  float out = 0.f;
  if (tid == 0) {
    int nw = (blockDim.x + WAVE - 1) / WAVE;
    for (int w = 0; w < nw; ++w) out += red[w];
    red[15] = out;
  }
  __syncthreads();
  return red[15];
}

// ---------------------------------------------------------------------------
// forward: y = x * w * rsqrt(mean(x^2) + eps); also emits inv_rms per row
// ---------------------------------------------------------------------------
__global__ void rmsnorm_fwd_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w,
    bf16* __restrict__ y, float* __restrict__ inv_rms,
    int rows, int H, float eps) {
  int r = blockIdx.x;
  if (r >= rows) return;
  const bf16* xr = x + (long)r * H;
  bf16* yr = y + (long)r * H;
  __shared__ float red[16];

  float ss = 0.f;
  for (int j = threadIdx.x * 8; j < H; j += blockDim.x * 8) {
    // 8 bf16 = 16 B per lane (vectorized load)
    const ushort4* p = reinterpret_cast<const ushort4*>(xr + j);
    ushort4 a = p[0], b = p[1];
    float v0 = bf2f(*reinterpret_cast<bf16*>(&a.x)), v1 = bf2f(*reinterpret_cast<bf16*>(&a.y));
    float v2 = bf2f(*reinterpret_cast<bf16*>(&a.z)), v3 = bf2f(*reinterpret_cast<bf16*>(&a.w));
    float v4 = bf2f(*reinterpret_cast<bf16*>(&b.x)), v5 = bf2f(*reinterpret_cast<bf16*>(&b.y));
    float v6 = bf2f(*reinterpret_cast<bf16*>(&b.z)), v7 = bf2f(*reinterpret_cast<bf16*>(&b.w));
    ss += v0*v0 + v1*v1 + v2*v2 + v3*v3 + v4*v4 + v5*v5 + v6*v6 + v7*v7;
  }
  float total = block_sum(ss, red);
  float ir = rsqrtf(total / H + eps);
  if (threadIdx.x == 0) inv_rms[r] = ir;

  for (int j = threadIdx.x; j < H; j += blockDim.x) {
    yr[j] = f2bf(bf2f(xr[j]) * ir * bf2f(w[j]));
  }
}

// ---------------------------------------------------------------------------
// backward: dx = ir * (dy*w) - x * ir^3/H * sum(dy*w*x); dw += dy * x * ir
// dw accumulated in fp32 via atomics (rows >> H contention is fine).
// ---------------------------------------------------------------------------
__global__ void rmsnorm_bwd_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w,
    const bf16* __restrict__ dy, const float* __restrict__ inv_rms,
    bf16* __restrict__ dx, float* __restrict__ dw,
    int rows, int H) {
  int r = blockIdx.x;
  if (r >= rows) return;
  const bf16* xr = x + (long)r * H;
  const bf16* dyr = dy + (long)r * H;
  bf16* dxr = dx + (long)r * H;
  float ir = inv_rms[r];
  __shared__ float red[16];

  float dot = 0.f;
  for (int j = threadIdx.x; j < H; j += blockDim.x) {
    dot += bf2f(dyr[j]) * bf2f(w[j]) * bf2f(xr[j]);
  }
  float total = block_sum(dot, red);
  float c = total * ir * ir * ir / H;

  for (int j = threadIdx.x; j < H; j += blockDim.x) {
    float xv = bf2f(xr[j]);
    float dyv = bf2f(dyr[j]);
    dxr[j] = f2bf(dyv * bf2f(w[j]) * ir - xv * c);
    atomicAdd(&dw[j], dyv * xv * ir);
  }
}

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  auto x2 = x.contiguous();
  int H = x2.size(-1);
  long rows = x2.numel() / H;
  TORCH_CHECK(H % 8 == 0, "rmsnorm: hidden must be a multiple of 8");
  auto y = torch::empty_like(x2);
  auto inv_rms = torch::empty({rows}, x2.options().dtype(torch::kFloat));
  int block = std::min(1024, ((H / 8 + WAVE - 1) / WAVE) * WAVE);
  block = std::max(block, WAVE);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(rmsnorm_fwd_kernel, dim3((int)rows), dim3(block), 0, stream,
      reinterpret_cast<const bf16*>(x2.data_ptr()),
      reinterpret_cast<const bf16*>(w.contiguous().data_ptr()),
      reinterpret_cast<bf16*>(y.data_ptr()), inv_rms.data_ptr<float>(),
      (int)rows, H, (float)eps);
  return {y, inv_rms};
}

std::vector<torch::Tensor> rmsnorm_bwd(
    torch::Tensor x, torch::Tensor w, torch::Tensor dy, torch::Tensor inv_rms) {
  auto x2 = x.contiguous();
  int H = x2.size(-1);
  long rows = x2.numel() / H;
  auto dx = torch::empty_like(x2);
  auto dw = torch::zeros({H}, x2.options().dtype(torch::kFloat));
  int block = std::min(1024, ((H + WAVE - 1) / WAVE) * WAVE);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(rmsnorm_bwd_kernel, dim3((int)rows), dim3(block), 0, stream,
      reinterpret_cast<const bf16*>(x2.data_ptr()),
      reinterpret_cast<const bf16*>(w.contiguous().data_ptr()),
      reinterpret_cast<const bf16*>(dy.contiguous().data_ptr()),
      inv_rms.data_ptr<float>(),
      reinterpret_cast<bf16*>(dx.data_ptr()), dw.data_ptr<float>(),
      (int)rows, H);
  return {dx, dw};
}

void init_norm_ops(pybind11::module_& m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd, "fused RMSNorm forward (bf16)");
  m.def("rmsnorm_bwd", &rmsnorm_bwd, "fused RMSNorm backward (bf16)");
}
