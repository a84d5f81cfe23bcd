// CDNA4 (gfx950 / MI355X) kernels for the RL hot paths.
//
// Replaces the CPU/numpy paths the reference framework uses for these ops
// (GAE scan rollout_buffer.py:472, n-step
// replay_buffer.py:287, C51 projection dqn_rainbow.py:389, polyak
// algo_utils.py:128, NoisyLinear custom_components.py:92, GRPO group
// advantage grpo.py:1219).
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
// - wave width 64; all block sizes are multiples of 64
// - memory-bound kernels use grid-stride loops capped near 2048 blocks
// - float4 vectorization where layout permits (G13)
// - no inter-workgroup ordering assumptions (G16)

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <vector>

#define WAVE 64
#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be a GPU tensor")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

static inline int grid_1d(long n, int block) {
  long g = (n + block - 1) / block;
  return (int)std::min(g, (long)2048 * 8);
}

// ---------------------------------------------------------------------------
// GAE reverse scan: one lane per env column, T-loop in-register.
// rewards/values/dones: (T, N) row-major; advantages out: (T, N).
// dones[t] marks termination AFTER step t (the collectors' convention), so
// step t's bootstrap through V(s_{t+1}) and its lambda carry are both cut by
// dones[t] itself.
// ---------------------------------------------------------------------------
__global__ void gae_scan_kernel(
    const float* __restrict__ rewards,
    const float* __restrict__ values,
    const float* __restrict__ dones,
    const float* __restrict__ last_value,
    float* __restrict__ adv,
    int T, int N, float gamma, float lam) {
  int n = blockIdx.x * blockDim.x + threadIdx.x;
  if (n >= N) return;
  float next_adv = 0.f;
  float next_val = last_value[n];
  for (int t = T - 1; t >= 0; --t) {
    long i = (long)t * N + n;
    float nd = 1.f - dones[i];
    float delta = rewards[i] + gamma * next_val * nd - values[i];
    next_adv = delta + gamma * lam * nd * next_adv;
    adv[i] = next_adv;
    next_val = values[i];
  }
}

torch::Tensor gae_scan(
    torch::Tensor rewards, torch::Tensor values, torch::Tensor dones,
    torch::Tensor last_value,
    double gamma, double lam) {
  CHECK_GPU(rewards); CHECK_CONTIG(rewards);
  int T = rewards.size(0), N = rewards.size(1);
  auto adv = torch::empty_like(rewards);
  int block = 256;
  int grid = (N + block - 1) / block;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(gae_scan_kernel, dim3(grid), dim3(block), 0, stream,
      rewards.data_ptr<float>(), values.data_ptr<float>(), dones.data_ptr<float>(),
      last_value.data_ptr<float>(),
      adv.data_ptr<float>(), T, N, (float)gamma, (float)lam);
  return adv;
}

// ---------------------------------------------------------------------------
// n-step windows: one lane per sampled row; window loop in-register.
// rewards/dones: (B, n).  Emits (returns, effective_steps).
// ---------------------------------------------------------------------------
__global__ void nstep_scan_kernel(
    const float* __restrict__ rewards,
    const float* __restrict__ dones,
    float* __restrict__ returns,
    float* __restrict__ steps,
    int B, int n, float gamma) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  for (; b < B; b += gridDim.x * blockDim.x) {
    const float* r = rewards + (long)b * n;
    const float* d = dones + (long)b * n;
    float acc = 0.f, disc = 1.f;
    int k = 0;
    for (; k < n; ++k) {
      acc += disc * r[k];
      if (d[k] > 0.5f) { ++k; break; }
      disc *= gamma;
    }
    returns[b] = acc;
    steps[b] = (float)k;
  }
}

std::vector<torch::Tensor> nstep_scan(torch::Tensor rewards, torch::Tensor dones, double gamma) {
  CHECK_GPU(rewards); CHECK_CONTIG(rewards);
  int B = rewards.size(0), n = rewards.size(1);
  auto returns = torch::empty({B}, rewards.options());
  auto steps = torch::empty({B}, rewards.options());
  int block = 256;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(nstep_scan_kernel, dim3(grid_1d(B, block)), dim3(block), 0, stream,
      rewards.data_ptr<float>(), dones.data_ptr<float>(),
      returns.data_ptr<float>(), steps.data_ptr<float>(), B, n, (float)gamma);
  return {returns, steps};
}

// ---------------------------------------------------------------------------
// C51 categorical projection. One thread per (row, atom); the two-sided
// scatter uses device-scope atomics (cross-lane rows never alias, atoms of
// one row do).  A (num_atoms) is small (51) so a whole row fits one wave.
// ---------------------------------------------------------------------------
__global__ void c51_project_kernel(
    const float* __restrict__ next_dist,  // (B, A)
    const float* __restrict__ rewards,    // (B,)
    const float* __restrict__ dones,      // (B,)
    const float* __restrict__ support,    // (A,)
    float* __restrict__ proj,             // (B, A) pre-zeroed
    int B, int A, float gamma, float v_min, float v_max, float inv_dz) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long total = (long)B * A;
  for (; i < total; i += (long)gridDim.x * blockDim.x) {
    int b = i / A, j = i % A;
    float tz = rewards[b] + (1.f - dones[b]) * gamma * support[j];
    tz = fminf(fmaxf(tz, v_min), v_max);
    float pos = (tz - v_min) * inv_dz;
    int lo = (int)floorf(pos);
    int up = (int)ceilf(pos);
    if (up == lo) {           // integral position: keep mass in one atom
      if (lo > 0) lo -= 1; else up += 1;
    }
    float p = next_dist[i];
    atomicAdd(&proj[(long)b * A + lo], p * ((float)up - pos));
    atomicAdd(&proj[(long)b * A + min(up, A - 1)], p * (pos - (float)lo));
  }
}

torch::Tensor c51_project(
    torch::Tensor next_dist, torch::Tensor rewards, torch::Tensor dones,
    torch::Tensor support, double gamma, double v_min, double v_max) {
  CHECK_GPU(next_dist); CHECK_CONTIG(next_dist);
  int B = next_dist.size(0), A = next_dist.size(1);
  auto proj = torch::zeros_like(next_dist);
  float dz = ((float)v_max - (float)v_min) / (A - 1);
  int block = 256;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(c51_project_kernel, dim3(grid_1d((long)B * A, block)), dim3(block), 0, stream,
      next_dist.data_ptr<float>(), rewards.data_ptr<float>(), dones.data_ptr<float>(),
      support.data_ptr<float>(), proj.data_ptr<float>(),
      B, A, (float)gamma, (float)v_min, (float)v_max, 1.f / dz);
  return proj;
}

// ---------------------------------------------------------------------------
// Fused polyak soft-update across a whole parameter set: one launch per
// network instead of one lerp per tensor.  Pointer table staged per call.
// ---------------------------------------------------------------------------
struct PtrPair { float* tgt; const float* src; long numel; long offset; };

__global__ void polyak_kernel(
    float** __restrict__ tgts, float** __restrict__ srcs,
    const long* __restrict__ numels, const long* __restrict__ offsets,
    int n_tensors, long total, float tau) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < total; i += (long)gridDim.x * blockDim.x) {
    // binary search for tensor containing flat index i
    int lo = 0, hi = n_tensors - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (offsets[mid] <= i) lo = mid; else hi = mid - 1;
    }
    long k = i - offsets[lo];
    float* t = tgts[lo];
    const float* s = srcs[lo];
    t[k] += tau * (s[k] - t[k]);
  }
}

void polyak_(std::vector<torch::Tensor> targets, std::vector<torch::Tensor> sources, double tau) {
  int n = targets.size();
  TORCH_CHECK(n == (int)sources.size(), "polyak_: length mismatch");
  if (n == 0) return;
  std::vector<float*> h_tgts(n);
  std::vector<float*> h_srcs(n);
  std::vector<long> h_numels(n), h_offsets(n);
  long total = 0;
  for (int i = 0; i < n; ++i) {
    TORCH_CHECK(targets[i].is_cuda() && targets[i].is_contiguous(), "polyak_: bad target");
    TORCH_CHECK(targets[i].scalar_type() == torch::kFloat, "polyak_: fp32 only");
    h_tgts[i] = targets[i].data_ptr<float>();
    h_srcs[i] = sources[i].data_ptr<float>();
    h_numels[i] = targets[i].numel();
    h_offsets[i] = total;
    total += h_numels[i];
  }
  auto opts = torch::TensorOptions().dtype(torch::kLong).device(targets[0].device());
  // stage the pointer table via one pinned H2D copy
  auto tgt_t = torch::from_blob(h_tgts.data(), {n}, torch::kLong).to(opts.device(), /*non_blocking=*/false);
  auto src_t = torch::from_blob(h_srcs.data(), {n}, torch::kLong).to(opts.device());
  auto numel_t = torch::from_blob(h_numels.data(), {n}, torch::kLong).to(opts.device());
  auto off_t = torch::from_blob(h_offsets.data(), {n}, torch::kLong).to(opts.device());
  int block = 256;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(polyak_kernel, dim3(grid_1d(total, block)), dim3(block), 0, stream,
      reinterpret_cast<float**>(tgt_t.data_ptr<long>()),
      reinterpret_cast<float**>(src_t.data_ptr<long>()),
      numel_t.data_ptr<long>(), off_t.data_ptr<long>(),
      n, total, (float)tau);
}

// ---------------------------------------------------------------------------
// NoisyLinear forward: fuse W_eff = mu + sigma*eps (and bias) in one pass,
// then the GEMM runs on rocBLAS via torch::mm.  Returns (out, w_eff) so the
// backward reuses W_eff without recomputation.
// ---------------------------------------------------------------------------
__global__ void noisy_weight_kernel(
    const float* __restrict__ mu, const float* __restrict__ sigma,
    const float* __restrict__ eps, float* __restrict__ out, long n) {
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long stride = (long)gridDim.x * blockDim.x * 4;
  for (; i + 3 < n; i += stride) {
    float4 m = *reinterpret_cast<const float4*>(mu + i);
    float4 s = *reinterpret_cast<const float4*>(sigma + i);
    float4 e = *reinterpret_cast<const float4*>(eps + i);
    float4 o;
    o.x = m.x + s.x * e.x; o.y = m.y + s.y * e.y;
    o.z = m.z + s.z * e.z; o.w = m.w + s.w * e.w;
    *reinterpret_cast<float4*>(out + i) = o;
  }
  // tail
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    for (long k = (n / 4) * 4; k < n; ++k) out[k] = mu[k] + sigma[k] * eps[k];
  }
}

static torch::Tensor fused_noisy_weight(torch::Tensor mu, torch::Tensor sigma, torch::Tensor eps) {
  auto out = torch::empty_like(mu);
  long n = mu.numel();
  int block = 256;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(noisy_weight_kernel, dim3(grid_1d((n + 3) / 4, block)), dim3(block), 0, stream,
      mu.data_ptr<float>(), sigma.data_ptr<float>(), eps.data_ptr<float>(),
      out.data_ptr<float>(), n);
  return out;
}

std::vector<torch::Tensor> noisy_linear_fwd(
    torch::Tensor x, torch::Tensor w_mu, torch::Tensor w_sigma, torch::Tensor w_eps,
    torch::Tensor b_mu, torch::Tensor b_sigma, torch::Tensor b_eps) {
  CHECK_GPU(x);
  auto w_eff = fused_noisy_weight(w_mu.contiguous(), w_sigma.contiguous(), w_eps.contiguous());
  auto b_eff = fused_noisy_weight(b_mu.contiguous(), b_sigma.contiguous(), b_eps.contiguous());
  auto x2 = x.reshape({-1, x.size(-1)});
  auto out = torch::addmm(b_eff, x2, w_eff.t());
  std::vector<int64_t> shape(x.sizes().begin(), x.sizes().end());
  shape.back() = w_mu.size(0);
  return {out.reshape(shape), w_eff};
}

// ---------------------------------------------------------------------------
// GRPO group-relative advantage: one wave per group (group_size <= a few
// hundred); shuffle-reduce mean and variance in-register.
// ---------------------------------------------------------------------------
__global__ void group_advantage_kernel(
    const float* __restrict__ rewards, float* __restrict__ adv,
    int n_groups, int G, bool scale, float eps) {
  int g = blockIdx.x;           // one block (one wave) per group
  if (g >= n_groups) return;
  int lane = threadIdx.x;       // blockDim.x == 64
  float sum = 0.f, sumsq = 0.f;
  for (int k = lane; k < G; k += WAVE) {
    float r = rewards[(long)g * G + k];
    sum += r; sumsq += r * r;
  }
  #pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    sum += __shfl_down(sum, off);
    sumsq += __shfl_down(sumsq, off);
  }
  sum = __shfl(sum, 0);
  sumsq = __shfl(sumsq, 0);
  float mean = sum / G;
  float var = fmaxf(sumsq / G - mean * mean, 0.f);
  // unbiased std to match torch.std (n-1 denominator)
  float std_u = (G > 1) ? sqrtf(var * G / (G - 1)) : 0.f;
  float inv = scale ? 1.f / (std_u + eps) : 1.f;
  for (int k = lane; k < G; k += WAVE) {
    long i = (long)g * G + k;
    adv[i] = (rewards[i] - mean) * inv;
  }
}

torch::Tensor group_advantage(torch::Tensor rewards, long group_size, bool scale, double eps) {
  CHECK_GPU(rewards); CHECK_CONTIG(rewards);
  long B = rewards.numel();
  TORCH_CHECK(B % group_size == 0, "rewards not divisible by group_size");
  int n_groups = B / group_size;
  auto adv = torch::empty_like(rewards);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(group_advantage_kernel, dim3(n_groups), dim3(WAVE), 0, stream,
      rewards.data_ptr<float>(), adv.data_ptr<float>(),
      n_groups, (int)group_size, scale, (float)eps);
  return adv;
}

// ---------------------------------------------------------------------------
// PER segment trees (reference CPU counterpart
// agilerl/components/segment_tree.py:119-222).  Tree layout: flat
// (2*capacity) array, root at 1, leaves at [capacity, 2*capacity).
//
// update: one workgroup, barrier per level — every thread scatters its
// leaves, then recomputes the ancestors of its leaves level by level.
// Shared ancestors are recomputed redundantly by several threads, but each
// write is a full recompute from already-final children, so racing writes
// store identical bits (benign).  Duplicate leaf indices resolve
// nondeterministically (same as the torch scatter path).
//
// descent: one lane per prefix value; the top levels of the tree are
// staged into LDS once per workgroup (32 KB = top 13 levels) so the
// latency-bound top of every descent hits LDS instead of HBM.
// ---------------------------------------------------------------------------

#define SEG_OP_SUM 0
#define SEG_OP_MIN 1
#define SEG_LDS_N 8192  // staged floats (32 KB LDS): nodes [0, 8192)

__global__ void segtree_update_kernel(
    float* __restrict__ tree,
    const long* __restrict__ idx,     // (B,) leaf indices in [0, capacity)
    const float* __restrict__ values, // (B,)
    long B, int capacity, int depth, int op) {
  int tid = threadIdx.x;
  int nthreads = blockDim.x;
  // scatter leaves
  for (long b = tid; b < B; b += nthreads) {
    tree[capacity + idx[b]] = values[b];
  }
  __syncthreads();
  // propagate: level k recomputes each updated leaf's ancestor at that level
  for (int k = 1; k <= depth; ++k) {
    for (long b = tid; b < B; b += nthreads) {
      long node = (capacity + idx[b]) >> k;
      float l = tree[2 * node], r = tree[2 * node + 1];
      tree[node] = (op == SEG_OP_MIN) ? fminf(l, r) : (l + r);
    }
    __syncthreads();
  }
}

void segtree_update(torch::Tensor tree, torch::Tensor idx, torch::Tensor values, long op) {
  CHECK_GPU(tree); CHECK_CONTIG(tree);
  CHECK_GPU(idx); CHECK_CONTIG(idx);
  int capacity = tree.size(0) / 2;
  int depth = 0;  // log2(capacity): leaf node (capacity+i) >> depth == 1
  while ((1 << depth) < capacity) ++depth;
  long B = idx.numel();
  if (B == 0) return;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(segtree_update_kernel, dim3(1), dim3(1024), 0, stream,
      tree.data_ptr<float>(), idx.data_ptr<long>(), values.data_ptr<float>(),
      B, capacity, depth, (int)op);
}

__device__ inline long segtree_descend_one(
    const float* __restrict__ tree, const float* __restrict__ lds,
    int lds_n, float p, int capacity, int depth) {
  long node = 1;
  for (int d = 0; d < depth; ++d) {
    long left = 2 * node;
    float lv = (left + 1 < lds_n) ? lds[left] : tree[left];
    if (p >= lv) { p -= lv; node = left + 1; } else { node = left; }
  }
  return node - capacity;
}

__global__ void segtree_retrieve_kernel(
    const float* __restrict__ tree,
    const float* __restrict__ prefix,
    long* __restrict__ out_idx,
    long B, int capacity, int depth) {
  __shared__ float lds[SEG_LDS_N];
  int lds_n = min(2 * capacity, SEG_LDS_N);
  for (int i = threadIdx.x; i < lds_n; i += blockDim.x) lds[i] = tree[i];
  __syncthreads();
  long b = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; b < B; b += (long)gridDim.x * blockDim.x) {
    out_idx[b] = segtree_descend_one(tree, lds, lds_n, prefix[b], capacity, depth);
  }
}

torch::Tensor segtree_retrieve(torch::Tensor tree, torch::Tensor prefix) {
  CHECK_GPU(tree); CHECK_CONTIG(tree); CHECK_GPU(prefix); CHECK_CONTIG(prefix);
  int capacity = tree.size(0) / 2;
  int depth = 0;
  while ((1 << depth) < capacity) ++depth;
  long B = prefix.numel();
  auto out = torch::empty({B}, prefix.options().dtype(torch::kLong));
  if (B == 0) return out;
  int block = 256;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(segtree_retrieve_kernel, dim3(grid_1d(B, block)), dim3(block), 0, stream,
      tree.data_ptr<float>(), prefix.data_ptr<float>(), out.data_ptr<long>(),
      B, capacity, depth);
  return out;
}

// Fused PER sample: stratified prefixes from uniform jitter, LDS descent,
// leaf gather, IS-weight computation — one launch, zero host syncs
// (total and p_min are read from the tree roots on device).
__global__ void per_sample_kernel(
    const float* __restrict__ sum_tree,
    const float* __restrict__ min_tree,
    const float* __restrict__ rand01,  // (B,) uniform jitter
    long* __restrict__ out_idx,
    float* __restrict__ out_w,
    long B, int capacity, int depth, long size, float beta) {
  __shared__ float lds[SEG_LDS_N];
  int lds_n = min(2 * capacity, SEG_LDS_N);
  for (int i = threadIdx.x; i < lds_n; i += blockDim.x) lds[i] = sum_tree[i];
  __syncthreads();
  float total = fmaxf(sum_tree[1], 1e-12f);
  float p_min = min_tree[1] / total;
  float max_w = (p_min > 0.f) ? powf(p_min * (float)size, -beta) : 1.f;
  long b = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; b < B; b += (long)gridDim.x * blockDim.x) {
    float u = ((float)b + rand01[b]) / (float)B * total;  // stratified
    long leaf = segtree_descend_one(sum_tree, lds, lds_n, u, capacity, depth);
    if (leaf > size - 1) leaf = size - 1;
    out_idx[b] = leaf;
    float p = sum_tree[capacity + leaf] / total;
    float pw = fmaxf(p * (float)size, 1e-12f);
    out_w[b] = powf(pw, -beta) / max_w;
  }
}

std::vector<torch::Tensor> per_sample(
    torch::Tensor sum_tree, torch::Tensor min_tree, torch::Tensor rand01,
    long size, double beta) {
  CHECK_GPU(sum_tree); CHECK_CONTIG(sum_tree);
  int capacity = sum_tree.size(0) / 2;
  int depth = 0;
  while ((1 << depth) < capacity) ++depth;
  long B = rand01.numel();
  auto idx = torch::empty({B}, rand01.options().dtype(torch::kLong));
  auto w = torch::empty({B}, rand01.options());
  int block = 256;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(per_sample_kernel, dim3(grid_1d(B, block)), dim3(block), 0, stream,
      sum_tree.data_ptr<float>(), min_tree.data_ptr<float>(),
      rand01.data_ptr<float>(), idx.data_ptr<long>(), w.data_ptr<float>(),
      B, capacity, depth, size, (float)beta);
  return {idx, w};
}

// ---------------------------------------------------------------------------

void init_lm_ops(pybind11::module_& m);    // lm_ops.hip
void init_norm_ops(pybind11::module_& m);  // norm_ops.hip
void init_act_ops(pybind11::module_& m);   // act_ops.hip
void init_paged_attn(pybind11::module_& m);  // paged_attn.hip
void init_skinny_gemm(pybind11::module_& m);  // skinny_gemm.hip

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  init_lm_ops(m);
  init_norm_ops(m);
  init_act_ops(m);
  init_paged_attn(m);
  init_skinny_gemm(m);
  m.def("gae_scan", &gae_scan, "GAE reverse scan (T,N)");
  m.def("nstep_scan", &nstep_scan, "n-step returns over sampled windows");
  m.def("c51_project", &c51_project, "C51 categorical projection");
  m.def("polyak_", &polyak_, "fused multi-tensor polyak update");
  m.def("noisy_linear_fwd", &noisy_linear_fwd, "fused NoisyLinear forward");
  m.def("group_advantage", &group_advantage, "GRPO group-relative advantage");
  m.def("segtree_update", &segtree_update, "PER segment-tree batched update+propagate");
  m.def("segtree_retrieve", &segtree_retrieve, "PER sum-tree prefix descent (LDS-staged)");
  m.def("per_sample", &per_sample, "fused PER stratified sample + IS weights");
}
