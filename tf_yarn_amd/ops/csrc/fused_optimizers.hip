// Fused optimizer update kernels for MI355X (gfx950).
//
// These replace the per-step optimizer apply the reference delegates to TF /
// torch (optimizer-level allreduce hook + server-side
// apply): one kernel per flat tensor (or per reducer bucket), fp32 states,
// grads in fp32 or bf16, with an optional bf16 "compute copy" of the params
// written in the same pass (mixed-precision master-weight training).
//
// All kernels are memory-bound elementwise: 16 B/lane f32x4 vector accesses,
// grid-stride, grid capped at 2048 blocks (guide Appendix B / G11/G13).

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

struct SgdArgs {
  float lr, momentum, dampening, weight_decay, scale;
  bool nesterov, first_step;
};

template <typename GIo>
__global__ void sgd_kernel(float* __restrict__ p,
                           const typename GIo::scalar_t* __restrict__ g,
                           float* __restrict__ mom,  // may be null
                           unsigned short* __restrict__ p_bf16,  // may be null
                           int64_t n, SgdArgs a) {
  const int64_t tid = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  const int64_t nvec = n >> 2;
  for (int64_t i = tid; i < nvec; i += stride) {
    f32x4 pv = reinterpret_cast<f32x4*>(p)[i];
    f32x4 mv;
    if (mom) mv = reinterpret_cast<f32x4*>(mom)[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gv = GIo::load(g, i * 4 + j) * a.scale;
      gv += a.weight_decay * pv[j];
      if (mom) {
        float m = a.first_step ? gv
                               : a.momentum * mv[j] + (1.f - a.dampening) * gv;
        mv[j] = m;
        gv = a.nesterov ? gv + a.momentum * m : m;
      }
      pv[j] -= a.lr * gv;
    }
    reinterpret_cast<f32x4*>(p)[i] = pv;
    if (mom) reinterpret_cast<f32x4*>(mom)[i] = mv;
    if (p_bf16) {
      bf16x4 bv;
#pragma unroll
      for (int j = 0; j < 4; ++j) bv[j] = f32_to_bf16(pv[j]);
      reinterpret_cast<bf16x4*>(p_bf16)[i] = bv;
    }
  }
  for (int64_t i = (nvec << 2) + tid; i < n; i += stride) {
    float pv = p[i];
    float gv = GIo::load(g, i) * a.scale;
    gv += a.weight_decay * pv;
    if (mom) {
      float m = a.first_step ? gv
                             : a.momentum * mom[i] + (1.f - a.dampening) * gv;
      mom[i] = m;
      gv = a.nesterov ? gv + a.momentum * m : m;
    }
    pv -= a.lr * gv;
    p[i] = pv;
    if (p_bf16) p_bf16[i] = f32_to_bf16(pv);
  }
}

struct AdamArgs {
  float lr, beta1, beta2, eps, weight_decay, bc1, bc2, scale;
  bool adamw;
};

template <typename GIo>
__global__ void adam_kernel(float* __restrict__ p,
                            const typename GIo::scalar_t* __restrict__ g,
                            float* __restrict__ m_, float* __restrict__ v_,
                            unsigned short* __restrict__ p_bf16,
                            int64_t n, AdamArgs a) {
  const int64_t tid = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  const int64_t nvec = n >> 2;
  for (int64_t i = tid; i < nvec; i += stride) {
    f32x4 pv = reinterpret_cast<f32x4*>(p)[i];
    f32x4 mv = reinterpret_cast<f32x4*>(m_)[i];
    f32x4 vv = reinterpret_cast<f32x4*>(v_)[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gv = GIo::load(g, i * 4 + j) * a.scale;
      if (a.adamw) pv[j] -= a.lr * a.weight_decay * pv[j];
      else gv += a.weight_decay * pv[j];
      mv[j] = a.beta1 * mv[j] + (1.f - a.beta1) * gv;
      vv[j] = a.beta2 * vv[j] + (1.f - a.beta2) * gv * gv;
      float mhat = mv[j] / a.bc1;
      float vhat = vv[j] / a.bc2;
      pv[j] -= a.lr * mhat / (sqrtf(vhat) + a.eps);
    }
    reinterpret_cast<f32x4*>(p)[i] = pv;
    reinterpret_cast<f32x4*>(m_)[i] = mv;
    reinterpret_cast<f32x4*>(v_)[i] = vv;
    if (p_bf16) {
      bf16x4 bv;
#pragma unroll
      for (int j = 0; j < 4; ++j) bv[j] = f32_to_bf16(pv[j]);
      reinterpret_cast<bf16x4*>(p_bf16)[i] = bv;
    }
  }
  for (int64_t i = (nvec << 2) + tid; i < n; i += stride) {
    float pv = p[i];
    float gv = GIo::load(g, i) * a.scale;
    if (a.adamw) pv -= a.lr * a.weight_decay * pv;
    else gv += a.weight_decay * pv;
    float m = a.beta1 * m_[i] + (1.f - a.beta1) * gv;
    float v = a.beta2 * v_[i] + (1.f - a.beta2) * gv * gv;
    m_[i] = m; v_[i] = v;
    pv -= a.lr * (m / a.bc1) / (sqrtf(v / a.bc2) + a.eps);
    p[i] = pv;
    if (p_bf16) p_bf16[i] = f32_to_bf16(pv);
  }
}

struct AdagradArgs { float lr, eps, weight_decay, scale; };

template <typename GIo>
__global__ void adagrad_kernel(float* __restrict__ p,
                               const typename GIo::scalar_t* __restrict__ g,
                               float* __restrict__ acc,
                               int64_t n, AdagradArgs a) {
  const int64_t tid = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  const int64_t nvec = n >> 2;
  for (int64_t i = tid; i < nvec; i += stride) {
    f32x4 pv = reinterpret_cast<f32x4*>(p)[i];
    f32x4 av = reinterpret_cast<f32x4*>(acc)[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gv = GIo::load(g, i * 4 + j) * a.scale;
      gv += a.weight_decay * pv[j];
      av[j] += gv * gv;
      pv[j] -= a.lr * gv / (sqrtf(av[j]) + a.eps);
    }
    reinterpret_cast<f32x4*>(p)[i] = pv;
    reinterpret_cast<f32x4*>(acc)[i] = av;
  }
  for (int64_t i = (nvec << 2) + tid; i < n; i += stride) {
    float gv = GIo::load(g, i) * a.scale;
    gv += a.weight_decay * p[i];
    acc[i] += gv * gv;
    p[i] -= a.lr * gv / (sqrtf(acc[i]) + a.eps);
  }
}

struct AdadeltaArgs { float lr, rho, eps, weight_decay, scale; };

template <typename GIo>
__global__ void adadelta_kernel(float* __restrict__ p,
                                const typename GIo::scalar_t* __restrict__ g,
                                float* __restrict__ sq,
                                float* __restrict__ acc_d,
                                int64_t n, AdadeltaArgs a) {
  const int64_t tid = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  const int64_t nvec = n >> 2;
  for (int64_t i = tid; i < nvec; i += stride) {
    f32x4 pv = reinterpret_cast<f32x4*>(p)[i];
    f32x4 sv = reinterpret_cast<f32x4*>(sq)[i];
    f32x4 dv = reinterpret_cast<f32x4*>(acc_d)[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gv = GIo::load(g, i * 4 + j) * a.scale;
      gv += a.weight_decay * pv[j];
      sv[j] = a.rho * sv[j] + (1.f - a.rho) * gv * gv;
      float dx = sqrtf(dv[j] + a.eps) / sqrtf(sv[j] + a.eps) * gv;
      dv[j] = a.rho * dv[j] + (1.f - a.rho) * dx * dx;
      pv[j] -= a.lr * dx;
    }
    reinterpret_cast<f32x4*>(p)[i] = pv;
    reinterpret_cast<f32x4*>(sq)[i] = sv;
    reinterpret_cast<f32x4*>(acc_d)[i] = dv;
  }
  for (int64_t i = (nvec << 2) + tid; i < n; i += stride) {
    float gv = GIo::load(g, i) * a.scale;
    gv += a.weight_decay * p[i];
    sq[i] = a.rho * sq[i] + (1.f - a.rho) * gv * gv;
    float dx = sqrtf(acc_d[i] + a.eps) / sqrtf(sq[i] + a.eps) * gv;
    acc_d[i] = a.rho * acc_d[i] + (1.f - a.rho) * dx * dx;
    p[i] -= a.lr * dx;
  }
}

// ---- multi-tensor SGD ------------------------------------------------------
// One launch for all dense params of a group (the per-param launches were
// ~10 x 4.6 us/step in the profile): gridDim.y = tensor index, each y-slice
// grid-strides its own tensor.  Kernel-arg struct holds up to MT_MAX
// tensors (HIP kernarg limit 4 KB).

#define MT_MAX 24

struct MtTensors {
  float* p[MT_MAX];
  const void* g[MT_MAX];
  float* m[MT_MAX];           // nullptr if no momentum
  unsigned short* pb[MT_MAX]; // nullptr if no bf16 copy
  long numel[MT_MAX];
  int n;
};

template <typename GIo>
__global__ void sgd_mt_kernel(MtTensors t, SgdArgs a) {
  const int ti = blockIdx.y;
  if (ti >= t.n) return;
  const int64_t n = t.numel[ti];
  float* __restrict__ p = t.p[ti];
  const typename GIo::scalar_t* __restrict__ g =
      reinterpret_cast<const typename GIo::scalar_t*>(t.g[ti]);
  float* __restrict__ mom = t.m[ti];
  unsigned short* __restrict__ pb = t.pb[ti];
  const int64_t tid = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  const int64_t nvec = n >> 2;
  for (int64_t i = tid; i < nvec; i += stride) {
    f32x4 pv = reinterpret_cast<f32x4*>(p)[i];
    f32x4 mv;
    if (mom) mv = reinterpret_cast<f32x4*>(mom)[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gv = GIo::load(g, i * 4 + j) * a.scale;
      gv += a.weight_decay * pv[j];
      if (mom) {
        float m = a.first_step ? gv
                               : a.momentum * mv[j] + (1.f - a.dampening) * gv;
        mv[j] = m;
        gv = a.nesterov ? gv + a.momentum * m : m;
      }
      pv[j] -= a.lr * gv;
    }
    reinterpret_cast<f32x4*>(p)[i] = pv;
    if (mom) reinterpret_cast<f32x4*>(mom)[i] = mv;
    if (pb) {
      bf16x4 bv;
#pragma unroll
      for (int j = 0; j < 4; ++j) bv[j] = f32_to_bf16(pv[j]);
      reinterpret_cast<bf16x4*>(pb)[i] = bv;
    }
  }
  for (int64_t i = (nvec << 2) + tid; i < n; i += stride) {
    float pv = p[i];
    float gv = GIo::load(g, i) * a.scale;
    gv += a.weight_decay * pv;
    if (mom) {
      float m = a.first_step ? gv
                             : a.momentum * mom[i] + (1.f - a.dampening) * gv;
      mom[i] = m;
      gv = a.nesterov ? gv + a.momentum * m : m;
    }
    pv -= a.lr * gv;
    p[i] = pv;
    if (pb) pb[i] = f32_to_bf16(pv);
  }
}

// ---- host-side checks/dispatch --------------------------------------------

void check_flat_f32(const torch::Tensor& t, const char* name, int64_t n) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
  TORCH_CHECK(t.numel() == n, name, " numel mismatch");
}

void check_grad(const torch::Tensor& g, int64_t n) {
  TORCH_CHECK(g.is_cuda() && g.is_contiguous(), "grad must be GPU+contig");
  TORCH_CHECK(g.scalar_type() == torch::kFloat32 ||
              g.scalar_type() == torch::kBFloat16,
              "grad must be fp32 or bf16");
  TORCH_CHECK(g.numel() == n, "grad numel mismatch");
}

unsigned short* bf16_copy_ptr(const c10::optional<torch::Tensor>& t,
                              int64_t n) {
  if (!t.has_value()) return nullptr;
  TORCH_CHECK(t->is_cuda() && t->is_contiguous() &&
              t->scalar_type() == torch::kBFloat16 && t->numel() == n,
              "param_bf16 must be a contiguous bf16 GPU tensor of same size");
  return reinterpret_cast<unsigned short*>(t->data_ptr());
}

}  // namespace

void fused_sgd(torch::Tensor param, torch::Tensor grad,
               c10::optional<torch::Tensor> momentum_buf,
               c10::optional<torch::Tensor> param_bf16,
               double lr, double momentum, double dampening,
               double weight_decay, bool nesterov, bool first_step,
               double grad_scale) {
  int64_t n = param.numel();
  check_flat_f32(param, "param", n);
  check_grad(grad, n);
  float* mom = nullptr;
  if (momentum_buf.has_value()) {
    check_flat_f32(*momentum_buf, "momentum", n);
    mom = momentum_buf->data_ptr<float>();
  }
  SgdArgs a{(float)lr, (float)momentum, (float)dampening,
            (float)weight_decay, (float)grad_scale, nesterov, first_step};
  auto stream = c10::hip::getCurrentHIPStream().stream();
  int grid = miyarn_grid((n + 3) / 4);
  if (grad.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(sgd_kernel<F32Io>, dim3(grid), dim3(MIYARN_BLOCK), 0,
                       stream, param.data_ptr<float>(),
                       grad.data_ptr<float>(), mom,
                       bf16_copy_ptr(param_bf16, n), n, a);
  } else {
    hipLaunchKernelGGL(sgd_kernel<Bf16Io>, dim3(grid), dim3(MIYARN_BLOCK), 0,
                       stream, param.data_ptr<float>(),
                       reinterpret_cast<unsigned short*>(grad.data_ptr()),
                       mom, bf16_copy_ptr(param_bf16, n), n, a);
  }
}

void fused_sgd_mt(std::vector<torch::Tensor> params,
                  std::vector<torch::Tensor> grads,
                  std::vector<torch::Tensor> momentum_bufs,  // may be empty
                  std::vector<torch::Tensor> params_bf16,    // may be empty
                  double lr, double momentum, double dampening,
                  double weight_decay, bool nesterov, bool first_step,
                  double grad_scale) {
  const int n = static_cast<int>(params.size());
  TORCH_CHECK(n > 0 && (int)grads.size() == n, "params/grads mismatch");
  const bool has_mom = !momentum_bufs.empty();
  const bool has_bf16 = !params_bf16.empty();
  TORCH_CHECK(!has_mom || (int)momentum_bufs.size() == n, "momentum size");
  TORCH_CHECK(!has_bf16 || (int)params_bf16.size() == n, "bf16 size");
  auto gtype = grads[0].scalar_type();
  SgdArgs a{(float)lr, (float)momentum, (float)dampening,
            (float)weight_decay, (float)grad_scale, nesterov, first_step};
  auto stream = c10::hip::getCurrentHIPStream().stream();
  for (int base = 0; base < n; base += MT_MAX) {
    MtTensors t{};
    int k = std::min(MT_MAX, n - base);
    t.n = k;
    int64_t max_numel = 0;
    for (int i = 0; i < k; ++i) {
      const auto& p = params[base + i];
      const auto& g = grads[base + i];
      int64_t ne = p.numel();
      check_flat_f32(p, "param", ne);
      check_grad(g, ne);
      TORCH_CHECK(g.scalar_type() == gtype,
                  "all grads in one fused_sgd_mt call share a dtype");
      t.p[i] = p.data_ptr<float>();
      t.g[i] = g.data_ptr();
      t.m[i] = nullptr;
      t.pb[i] = nullptr;
      if (has_mom) {
        check_flat_f32(momentum_bufs[base + i], "momentum", ne);
        t.m[i] = momentum_bufs[base + i].data_ptr<float>();
      }
      if (has_bf16) {
        t.pb[i] = bf16_copy_ptr(
            c10::optional<torch::Tensor>(params_bf16[base + i]), ne);
      }
      t.numel[i] = ne;
      max_numel = std::max(max_numel, ne);
    }
    dim3 grid(miyarn_grid((max_numel + 3) / 4), k);
    if (gtype == torch::kFloat32) {
      hipLaunchKernelGGL(sgd_mt_kernel<F32Io>, grid, dim3(MIYARN_BLOCK), 0,
                         stream, t, a);
    } else {
      hipLaunchKernelGGL(sgd_mt_kernel<Bf16Io>, grid, dim3(MIYARN_BLOCK), 0,
                         stream, t, a);
    }
  }
}

void fused_adam(torch::Tensor param, torch::Tensor grad,
                torch::Tensor exp_avg, torch::Tensor exp_avg_sq,
                c10::optional<torch::Tensor> param_bf16,
                double lr, double beta1, double beta2, double eps,
                double weight_decay, bool adamw, int64_t step,
                double grad_scale) {
  int64_t n = param.numel();
  check_flat_f32(param, "param", n);
  check_flat_f32(exp_avg, "exp_avg", n);
  check_flat_f32(exp_avg_sq, "exp_avg_sq", n);
  check_grad(grad, n);
  AdamArgs a{(float)lr, (float)beta1, (float)beta2, (float)eps,
             (float)weight_decay,
             (float)(1.0 - std::pow(beta1, (double)step)),
             (float)(1.0 - std::pow(beta2, (double)step)),
             (float)grad_scale, adamw};
  auto stream = c10::hip::getCurrentHIPStream().stream();
  int grid = miyarn_grid((n + 3) / 4);
  if (grad.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(adam_kernel<F32Io>, dim3(grid), dim3(MIYARN_BLOCK), 0,
                       stream, param.data_ptr<float>(),
                       grad.data_ptr<float>(), exp_avg.data_ptr<float>(),
                       exp_avg_sq.data_ptr<float>(),
                       bf16_copy_ptr(param_bf16, n), n, a);
  } else {
    hipLaunchKernelGGL(adam_kernel<Bf16Io>, dim3(grid), dim3(MIYARN_BLOCK), 0,
                       stream, param.data_ptr<float>(),
                       reinterpret_cast<unsigned short*>(grad.data_ptr()),
                       exp_avg.data_ptr<float>(),
                       exp_avg_sq.data_ptr<float>(),
                       bf16_copy_ptr(param_bf16, n), n, a);
  }
}

void fused_adagrad(torch::Tensor param, torch::Tensor grad,
                   torch::Tensor state_sum, double lr, double eps,
                   double weight_decay, double grad_scale) {
  int64_t n = param.numel();
  check_flat_f32(param, "param", n);
  check_flat_f32(state_sum, "state_sum", n);
  check_grad(grad, n);
  AdagradArgs a{(float)lr, (float)eps, (float)weight_decay,
                (float)grad_scale};
  auto stream = c10::hip::getCurrentHIPStream().stream();
  int grid = miyarn_grid((n + 3) / 4);
  if (grad.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(adagrad_kernel<F32Io>, dim3(grid), dim3(MIYARN_BLOCK),
                       0, stream, param.data_ptr<float>(),
                       grad.data_ptr<float>(), state_sum.data_ptr<float>(),
                       n, a);
  } else {
    hipLaunchKernelGGL(adagrad_kernel<Bf16Io>, dim3(grid), dim3(MIYARN_BLOCK),
                       0, stream, param.data_ptr<float>(),
                       reinterpret_cast<unsigned short*>(grad.data_ptr()),
                       state_sum.data_ptr<float>(), n, a);
  }
}

void fused_adadelta(torch::Tensor param, torch::Tensor grad,
                    torch::Tensor square_avg, torch::Tensor acc_delta,
                    double lr, double rho, double eps, double weight_decay,
                    double grad_scale) {
  int64_t n = param.numel();
  check_flat_f32(param, "param", n);
  check_flat_f32(square_avg, "square_avg", n);
  check_flat_f32(acc_delta, "acc_delta", n);
  check_grad(grad, n);
  AdadeltaArgs a{(float)lr, (float)rho, (float)eps, (float)weight_decay,
                 (float)grad_scale};
  auto stream = c10::hip::getCurrentHIPStream().stream();
  int grid = miyarn_grid((n + 3) / 4);
  if (grad.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(adadelta_kernel<F32Io>, dim3(grid), dim3(MIYARN_BLOCK),
                       0, stream, param.data_ptr<float>(),
                       grad.data_ptr<float>(), square_avg.data_ptr<float>(),
                       acc_delta.data_ptr<float>(), n, a);
  } else {
    hipLaunchKernelGGL(adadelta_kernel<Bf16Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       param.data_ptr<float>(),
                       reinterpret_cast<unsigned short*>(grad.data_ptr()),
                       square_avg.data_ptr<float>(),
                       acc_delta.data_ptr<float>(), n, a);
  }
}
