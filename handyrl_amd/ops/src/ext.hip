// handyrl_amd CDNA4 (gfx950) kernels + torch bindings.
//
// Native HIP, written for MI355X: wave64, one launch per logical op.
// Components:
//   * target_scan     — fused backward scans for TD(lambda) / UPGO / V-Trace
//                       (semantics of reference handyrl/losses.py:20-60, one
//                       kernel instead of ~3*T tiny launches per call)
//   * masked_sample   — batched legal-action softmax sampling for GPU actors
//                       (semantics of reference handyrl/generation.py:53-60)

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>

#define DEV_INLINE __device__ __forceinline__

namespace {

enum ScanKind { KIND_TD = 0, KIND_UPGO = 1, KIND_VTRACE = 2 };

// One lane scans one (b, p) trajectory serially over T.  Work per call is
// tiny (B*P lanes x T steps); the win over eager is launch-count, not FLOPs.
__global__ void target_scan_kernel(
    const float* __restrict__ values,    // (B,T,P)
    const float* __restrict__ ret_boot,  // (B,P): bootstrap value (returns at T-1)
    const float* __restrict__ rewards,   // (B,T,P) or nullptr
    const float* __restrict__ lambda_,   // (B,T,P)
    const float* __restrict__ rhos,      // (B,T,P), VTRACE only
    const float* __restrict__ cs,        // (B,T,P), VTRACE only
    float* __restrict__ targets,
    float* __restrict__ adv,
    int n_bp, int T, int P, float gamma, int kind)
{
    const int idx = blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= n_bp) return;
    const int b = idx / P, p = idx % P;
    const long base = (long)b * T * P + p;
    const long stride = P;

#define AT(a, t) a[base + (long)(t) * stride]

    const float ret_last = ret_boot[(long)b * P + p];

    if (kind == KIND_VTRACE) {
        // vs(t) = V(t) + sum-scan of deltas; advantage uses vs(t+1)
        float carry = 0.f;           // vs_minus_v at t+1
        float vs_next = ret_last;    // vs at t+1 (bootstrap = final return)
        for (int t = T - 1; t >= 0; --t) {
            const float v = AT(values, t);
            const float r = rewards ? AT(rewards, t) : 0.f;
            const float v_next = (t == T - 1) ? ret_last : AT(values, t + 1);
            const float delta = AT(rhos, t) * (r + gamma * v_next - v);
            const float vmv = (t == T - 1)
                ? delta
                : delta + gamma * AT(lambda_, t + 1) * AT(cs, t) * carry;
            carry = vmv;
            const float vs = vmv + v;
            AT(targets, t) = vs;
            AT(adv, t) = r + gamma * vs_next - v;
            vs_next = vs;
        }
    } else {
        float tv = ret_last;
        AT(targets, T - 1) = tv;
        AT(adv, T - 1) = tv - AT(values, T - 1);
        for (int t = T - 2; t >= 0; --t) {
            const float v1 = AT(values, t + 1);
            const float r = rewards ? AT(rewards, t) : 0.f;
            const float lam = AT(lambda_, t + 1);
            float mix = (1.f - lam) * v1 + lam * tv;
            if (kind == KIND_UPGO) mix = fmaxf(v1, mix);
            tv = r + gamma * mix;
            AT(targets, t) = tv;
            AT(adv, t) = tv - AT(values, t);
        }
    }
#undef AT
}

// One lane per row; A <= a few hundred, all L2-resident.  Three passes in
// registers: row max, exp-sum, then inverse-CDF selection at u * sum.
__global__ void masked_sample_kernel(
    const float* __restrict__ logits,   // (N, A)
    const float* __restrict__ mask,     // (N, A), additive: 0 legal / 1e32 illegal
    const float* __restrict__ uniform,  // (N,) in [0, 1)
    long* __restrict__ action,
    float* __restrict__ prob,
    int N, int A)
{
    const int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= N) return;
    const float* lg = logits + (long)i * A;
    const float* mk = mask + (long)i * A;

    float m = -INFINITY;
    for (int a = 0; a < A; ++a) m = fmaxf(m, lg[a] - mk[a]);
    float s = 0.f;
    for (int a = 0; a < A; ++a) s += __expf(lg[a] - mk[a] - m);

    const float u = uniform[i] * s;
    float acc = 0.f;
    int sel = -1;
    float pr = 0.f;
    for (int a = 0; a < A; ++a) {
        const float e = __expf(lg[a] - mk[a] - m);
        acc += e;
        if (sel < 0 && u < acc) { sel = a; pr = e / s; }
    }
    if (sel < 0) {   // numerical tail: last legal action
        for (int a = A - 1; a >= 0; --a) {
            if (mk[a] == 0.f) {
                sel = a;
                pr = __expf(lg[a] - mk[a] - m) / s;
                break;
            }
        }
        if (sel < 0) sel = 0;
    }
    action[i] = sel;
    prob[i] = pr;
}

}  // namespace

static std::vector<torch::Tensor> target_scan(
    torch::Tensor values, torch::Tensor ret_boot,
    c10::optional<torch::Tensor> rewards, torch::Tensor lambda_,
    c10::optional<torch::Tensor> rhos, c10::optional<torch::Tensor> cs,
    double gamma, int64_t kind)
{
    TORCH_CHECK(values.is_cuda() && values.scalar_type() == torch::kFloat32,
                "target_scan: values must be CUDA float32");
    TORCH_CHECK(values.dim() >= 3, "target_scan: expect (B,T,P,...) layout");
    const int B = values.size(0), T = values.size(1);
    int P = 1;
    for (int d = 2; d < values.dim(); ++d) P *= values.size(d);
    TORCH_CHECK(ret_boot.numel() == (long)B * P,
                "target_scan: bootstrap tensor must have B*P elements");

    auto targets = torch::empty_like(values);
    auto adv = torch::empty_like(values);
    const int n_bp = B * P;
    const int block = 256;
    const int grid = (n_bp + block - 1) / block;
    auto stream = at::cuda::getCurrentCUDAStream();

    const float* rew_ptr = rewards.has_value() ? rewards->data_ptr<float>() : nullptr;
    const float* rho_ptr = rhos.has_value() ? rhos->data_ptr<float>() : nullptr;
    const float* cs_ptr = cs.has_value() ? cs->data_ptr<float>() : nullptr;

    hipLaunchKernelGGL(target_scan_kernel, dim3(grid), dim3(block), 0, stream,
        values.data_ptr<float>(), ret_boot.data_ptr<float>(), rew_ptr,
        lambda_.data_ptr<float>(), rho_ptr, cs_ptr,
        targets.data_ptr<float>(), adv.data_ptr<float>(),
        n_bp, T, P, (float)gamma, (int)kind);
    return {targets, adv};
}

static std::vector<torch::Tensor> masked_sample(
    torch::Tensor logits, torch::Tensor mask, torch::Tensor uniform)
{
    TORCH_CHECK(logits.is_cuda() && logits.dim() == 2, "masked_sample: (N,A) CUDA expected");
    const int N = logits.size(0), A = logits.size(1);
    auto action = torch::empty({N}, logits.options().dtype(torch::kInt64));
    auto prob = torch::empty({N}, logits.options());
    const int block = 256;
    const int grid = (N + block - 1) / block;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(masked_sample_kernel, dim3(grid), dim3(block), 0, stream,
        logits.data_ptr<float>(), mask.data_ptr<float>(), uniform.data_ptr<float>(),
        action.data_ptr<long>(), prob.data_ptr<float>(), N, A);
    return {action, prob};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("target_scan", &target_scan,
          "fused TD/UPGO/V-Trace backward scan (targets, advantages)");
    m.def("masked_sample", &masked_sample,
          "batched masked-softmax action sampling (actions, probs)");
}
