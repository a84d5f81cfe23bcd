// CDNA4 (gfx950) kernels for the LLM fine-tuning path.
//
// fused linear logprobs (chunked lm_head GEMM +
// online logsumexp + target gather, fwd+bwd), fused token-masked
// GRPO/CISPO surrogate loss, masked reductions.  The chunked GEMM itself
// runs on rocBLAS/hipBLASLt (library GEMMs per the MI355X guide); the
// hand-written kernels here fuse everything AROUND the GEMM so the (N, V)
// logits chunk is consumed in one pass and never re-read from HBM.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <vector>

#define WAVE 64

// ---------------------------------------------------------------------------
// Row logsumexp + target gather over a (rows, V) logits chunk.
// One block (4 waves) per row: pass 1 max, pass 2 sum exp.  fp32 in/out.
// Emits logprob[r] = logits[r, tgt[r]] - logsumexp(logits[r, :]).
// Also stores lse[r] for the backward pass.
// ---------------------------------------------------------------------------
__global__ void row_lse_gather_kernel(
    const float* __restrict__ logits,   // (rows, V)
    const long* __restrict__ targets,   // (rows,)
    float* __restrict__ logprob,        // (rows,)
    float* __restrict__ lse,            // (rows,)
    int rows, long V, float inv_temp) {
  int r = blockIdx.x;
  if (r >= rows) return;
  const float* row = logits + (long)r * V;
  int tid = threadIdx.x;
  int nthreads = blockDim.x;
  int nwaves = nthreads / WAVE;
  int wid = tid / WAVE;
  int lane = tid & (WAVE - 1);
  __shared__ float red[16];
  __shared__ float bcast;

  // pass 1: max
  float m = -INFINITY;
  for (long j = tid; j < V; j += nthreads) m = fmaxf(m, row[j] * inv_temp);
  #pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) m = fmaxf(m, __shfl_down(m, off));
  if (lane == 0) red[wid] = m;
  __syncthreads();
  if (tid == 0) {
    float mm = red[0];
    for (int w = 1; w < nwaves; ++w) mm = fmaxf(mm, red[w]);
    bcast = mm;
  }
  __syncthreads();
  m = bcast;
  __syncthreads();  // red[] reused below

  // pass 2: sum of exp
  float s = 0.f;
  for (long j = tid; j < V; j += nthreads) s += __expf(row[j] * inv_temp - m);
  #pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) s += __shfl_down(s, off);
  if (lane == 0) red[wid] = s;
  __syncthreads();
  if (tid == 0) {
    float ss = 0.f;
    for (int w = 0; w < nwaves; ++w) ss += red[w];
    float l = m + __logf(ss);
    lse[r] = l;
    logprob[r] = row[targets[r]] * inv_temp - l;
  }
}

std::vector<torch::Tensor> row_lse_gather(
    torch::Tensor logits, torch::Tensor targets, double temperature) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous());
  int rows = logits.size(0);
  long V = logits.size(1);
  auto logprob = torch::empty({rows}, logits.options());
  auto lse = torch::empty({rows}, logits.options());
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(row_lse_gather_kernel, dim3(rows), dim3(256), 0, stream,
      logits.data_ptr<float>(), targets.data_ptr<long>(),
      logprob.data_ptr<float>(), lse.data_ptr<float>(),
      rows, V, (float)(1.0 / temperature));
  return {logprob, lse};
}

// ---------------------------------------------------------------------------
// Backward of the row logprob-gather: the forward emits
// logp = logits[t]*it - logsumexp(logits*it), so
// d(logp)/d(logits[j]) = it * (onehot[j==t] - softmax[j]).
// Computed in-place over the recomputed logits chunk (saves one (N,V)
// allocation + full re-read).  logits is overwritten with the gradient.
// ---------------------------------------------------------------------------
__global__ void row_softmax_bwd_kernel(
    float* __restrict__ logits,         // (rows, V) -> overwritten with grad
    const long* __restrict__ targets,
    const float* __restrict__ lse,
    const float* __restrict__ grad_logprob,  // (rows,)
    int rows, long V, float inv_temp) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long total = (long)rows * V;
  for (; i < total; i += (long)gridDim.x * blockDim.x) {
    int r = i / V;
    long j = i % V;
    float p = __expf(logits[i] * inv_temp - lse[r]);
    float g = ((j == targets[r] ? 1.f : 0.f) - p) * grad_logprob[r] * inv_temp;
    logits[i] = g;
  }
}

void row_softmax_bwd_(
    torch::Tensor logits, torch::Tensor targets, torch::Tensor lse,
    torch::Tensor grad_logprob, double temperature) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous());
  int rows = logits.size(0);
  long V = logits.size(1);
  long total = (long)rows * V;
  int block = 256;
  long g = std::min((total + block - 1) / block, (long)16384);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(row_softmax_bwd_kernel, dim3((int)g), dim3(block), 0, stream,
      logits.data_ptr<float>(), targets.data_ptr<long>(), lse.data_ptr<float>(),
      grad_logprob.data_ptr<float>(), rows, V, (float)(1.0 / temperature));
}

// ---------------------------------------------------------------------------
// Fused token-masked GRPO/CISPO surrogate: per token
//   ratio = exp(logp - old_logp)
//   clipped surrogate (GRPO) or clamped-IS (CISPO), optional k3 KL to ref
// One pass over (B*T) tokens, masked accumulation into per-sequence sums.
// Emits per-token loss grad d(loss)/d(logp) so the backward into the
// logprob kernel is a single multiply — no (B,T,V) materialization.
// ---------------------------------------------------------------------------
__global__ void grpo_token_loss_kernel(
    const float* __restrict__ logp,      // (N,) current policy
    const float* __restrict__ old_logp,  // (N,)
    const float* __restrict__ ref_logp,  // (N,) or nullptr
    const float* __restrict__ adv,       // (N,) advantage broadcast per token
    const float* __restrict__ mask,      // (N,)
    float* __restrict__ loss_tok,        // (N,) token loss (masked)
    float* __restrict__ dlogp,           // (N,) d(loss)/d(logp) (masked)
    long N, float clip_lo, float clip_hi, float kl_coef, int cispo) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < N; i += (long)gridDim.x * blockDim.x) {
    float mk = mask[i];
    if (mk == 0.f) { loss_tok[i] = 0.f; dlogp[i] = 0.f; continue; }
    float lr = logp[i] - old_logp[i];
    float ratio = __expf(lr);
    float a = adv[i];
    float loss, dl;
    if (cispo) {
      // CISPO: clamp(ratio) treated as constant weight, grad flows via logp
      float w = fminf(fmaxf(ratio, clip_lo), clip_hi);
      loss = -w * a * logp[i];
      dl = -w * a;
    } else {
      float un = ratio * a;
      float cl = fminf(fmaxf(ratio, clip_lo), clip_hi) * a;
      if (un <= cl) {       // surrogate = min(un, cl); loss = -min
        loss = -un;
        dl = -ratio * a;    // d(-ratio*a)/dlogp = -ratio*a
      } else {
        loss = -cl;
        bool clipped = (ratio < clip_lo) || (ratio > clip_hi);
        dl = clipped ? 0.f : -ratio * a;
      }
    }
    if (ref_logp != nullptr && kl_coef != 0.f) {
      // k3 KL: exp(ref-logp) - (ref-logp) - 1 ; d/dlogp = 1 - exp(ref-logp)
      float d = ref_logp[i] - logp[i];
      float ed = __expf(d);
      loss += kl_coef * (ed - d - 1.f);
      dl += kl_coef * (1.f - ed);
    }
    loss_tok[i] = loss * mk;
    dlogp[i] = dl * mk;
  }
}

std::vector<torch::Tensor> grpo_token_loss(
    torch::Tensor logp, torch::Tensor old_logp,
    c10::optional<torch::Tensor> ref_logp,
    torch::Tensor adv, torch::Tensor mask,
    double clip_lo, double clip_hi, double kl_coef, bool cispo) {
  TORCH_CHECK(logp.is_cuda() && logp.is_contiguous());
  long N = logp.numel();
  auto loss_tok = torch::empty_like(logp);
  auto dlogp = torch::empty_like(logp);
  const float* ref_ptr = ref_logp.has_value() ? ref_logp->data_ptr<float>() : nullptr;
  int block = 256;
  long g = std::min((N + block - 1) / block, (long)16384);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(grpo_token_loss_kernel, dim3((int)g), dim3(block), 0, stream,
      logp.data_ptr<float>(), old_logp.data_ptr<float>(), ref_ptr,
      adv.data_ptr<float>(), mask.data_ptr<float>(),
      loss_tok.data_ptr<float>(), dlogp.data_ptr<float>(),
      N, (float)clip_lo, (float)clip_hi, (float)kl_coef, cispo ? 1 : 0);
  return {loss_tok, dlogp};
}

// ---------------------------------------------------------------------------

void init_lm_ops(pybind11::module_& m) {
  m.def("row_lse_gather", &row_lse_gather, "row logsumexp + target gather");
  m.def("row_softmax_bwd_", &row_softmax_bwd_, "in-place softmax backward");
  m.def("grpo_token_loss", &grpo_token_loss, "fused token-masked GRPO/CISPO loss");
}
