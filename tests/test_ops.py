"""CPU-side tests for the op wrappers: the CPU reference implementations
must match stock torch optimizers/ops exactly (they are the baseline the
GPU numerics tests compare against)."""

import pytest
import torch
from torch import nn

from tf_yarn_amd import ops
from tf_yarn_amd.ops.optim import (FusedAdadelta, FusedAdagrad, FusedAdam,
                                   FusedSGD)


def _clone_params(model):
    return [p.detach().clone() for p in model.parameters()]


def _train(model, opt, steps=5, seed=7):
    torch.manual_seed(seed)
    xs = [torch.randn(8, 6) for _ in range(steps)]
    ys = [torch.randn(8, 3) for _ in range(steps)]
    for x, y in zip(xs, ys):
        opt.zero_grad()
        nn.functional.mse_loss(model(x), y).backward()
        opt.step()


@pytest.mark.parametrize("fused_cls,torch_cls,kwargs", [
    (FusedSGD, torch.optim.SGD, dict(lr=0.1)),
    (FusedSGD, torch.optim.SGD,
     dict(lr=0.1, momentum=0.9, weight_decay=0.01)),
    (FusedSGD, torch.optim.SGD,
     dict(lr=0.1, momentum=0.9, nesterov=True)),
    (FusedAdam, torch.optim.Adam,
     dict(lr=0.01, weight_decay=0.001)),
    (FusedAdagrad, torch.optim.Adagrad, dict(lr=0.05)),
    (FusedAdadelta, torch.optim.Adadelta, dict(lr=0.9)),
])
def test_fused_optimizer_matches_torch(fused_cls, torch_cls, kwargs):
    torch.manual_seed(0)
    m1 = nn.Sequential(nn.Linear(6, 16), nn.Tanh(), nn.Linear(16, 3))
    torch.manual_seed(0)
    m2 = nn.Sequential(nn.Linear(6, 16), nn.Tanh(), nn.Linear(16, 3))
    _train(m1, fused_cls(m1.parameters(), **kwargs))
    _train(m2, torch_cls(m2.parameters(), **kwargs))
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-5), \
            f"{fused_cls.__name__} diverges from {torch_cls.__name__}"


def test_fused_adamw_matches_torch():
    torch.manual_seed(0)
    m1 = nn.Linear(6, 3)
    torch.manual_seed(0)
    m2 = nn.Linear(6, 3)
    _train(m1, FusedAdam(m1.parameters(), lr=0.01, weight_decay=0.05,
                         adamw=True))
    _train(m2, torch.optim.AdamW(m2.parameters(), lr=0.01,
                                 weight_decay=0.05))
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-5)


def test_emb_fwd_cpu_matches_index_select():
    table = torch.randn(50, 8)
    ids = torch.randint(0, 50, (30,))
    out = ops.emb_fwd(table, ids)
    assert torch.equal(out, table.index_select(0, ids))
    out16 = ops.emb_fwd(table, ids, out_bf16=True)
    assert out16.dtype == torch.bfloat16


def test_emb_bwd_sgd_cpu():
    table = torch.randn(20, 4)
    table0 = table.clone()
    ids = torch.tensor([3, 3, 7])
    grad = torch.randn(3, 4)
    ops.emb_bwd_sgd(table, ids, grad, lr=0.5, scale=2.0)
    expected = table0.index_add(0, ids, grad, alpha=-1.0)
    assert torch.allclose(table, expected, atol=1e-6)


def test_bias_relu_autograd_matches_torch():
    x = torch.randn(10, 6, requires_grad=True)
    b = torch.randn(6, requires_grad=True)
    y = ops.bias_relu(x, b)
    loss = (y ** 2).sum()
    loss.backward()
    x2 = x.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    y2 = torch.relu(x2 + b2)
    ((y2 ** 2).sum()).backward()
    assert torch.allclose(y, y2)
    assert torch.allclose(x.grad, x2.grad)
    assert torch.allclose(b.grad, b2.grad)


def test_convert_scaled_cpu():
    src = torch.randn(13)
    dst = torch.empty(13, dtype=torch.bfloat16)
    ops.convert_scaled(src, dst, 2.0)
    assert torch.allclose(dst.float(), (src * 2).to(torch.bfloat16).float())


def test_col_reduce_dot_cpu():
    x = torch.randn(50, 8)
    dy = torch.randn(50)
    out = ops.col_reduce_dot(x, dy)
    ref = (x * dy.unsqueeze(1)).sum(dim=0)
    assert torch.allclose(out, ref, atol=1e-5)


def test_scalar_head_fn_matches_autograd():
    torch.manual_seed(4)
    x = torch.randn(32, 8, requires_grad=True)
    w = torch.randn(8, requires_grad=True)
    b = torch.zeros(1, requires_grad=True)
    y = ops.ScalarHeadFn.apply(x, w, b)
    y.pow(2).sum().backward()
    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    (x2 @ w2 + b2).pow(2).sum().backward()
    assert torch.allclose(y, x2 @ w2 + b2, atol=1e-6)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(w.grad, w2.grad, atol=1e-4)
    assert torch.allclose(b.grad, b2.grad, atol=1e-5)


def test_linear_bias_relu_matches_autograd():
    torch.manual_seed(5)
    x = torch.randn(16, 8, requires_grad=True)
    w = torch.randn(6, 8, requires_grad=True)
    b = torch.randn(6, requires_grad=True)
    y = ops.linear_bias_relu(x, w, b)
    y.pow(2).sum().backward()
    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    torch.relu(x2 @ w2.t() + b2).pow(2).sum().backward()
    assert torch.allclose(y, torch.relu(x @ w.t() + b).detach())
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(w.grad, w2.grad, atol=1e-5)
    assert torch.allclose(b.grad, b2.grad, atol=1e-5)


def test_gemm_bt_cpu_fallback():
    torch.manual_seed(23)
    a = torch.randn(32, 24)
    b = torch.randn(16, 24)
    bias = torch.randn(16)
    out = ops.gemm_bt(a, b, bias, relu=True)
    ref = torch.relu(a @ b.t() + bias)
    assert torch.allclose(out, ref, atol=1e-5)
    out2 = ops.gemm_bt(a, b, None, relu=False)
    assert torch.allclose(out2, a @ b.t(), atol=1e-5)


def test_row_dot_cpu_fallback():
    torch.manual_seed(30)
    x = torch.randn(64, 12)
    w = torch.randn(12)
    b = torch.randn(1)
    assert torch.allclose(ops.row_dot(x, w, b), x @ w + b, atol=1e-5)
    assert torch.allclose(ops.row_dot(x, w, None), x @ w, atol=1e-5)


def test_bce_head_loss_cpu_matches_torch():
    torch.manual_seed(31)
    parts = [torch.randn(40, requires_grad=True) for _ in range(3)]
    labels = (torch.rand(40) > 0.5).float()
    loss = ops.bce_head_loss(*parts, labels)
    ref_parts = [p.detach().clone().requires_grad_(True) for p in parts]
    ref = torch.nn.functional.binary_cross_entropy_with_logits(
        ref_parts[0] + ref_parts[1] + ref_parts[2], labels)
    assert torch.allclose(loss, ref, atol=1e-6)
    loss.backward()
    ref.backward()
    for p, rp in zip(parts, ref_parts):
        assert torch.allclose(p.grad, rp.grad, atol=1e-6)


def test_emb_fwd_into_and_gather_scatter_cpu():
    torch.manual_seed(32)
    table = torch.randn(20, 4)
    ids = torch.randint(0, 20, (6, 3))
    out = torch.zeros(6, 2 + 12)
    ops.emb_fwd_into(table, ids.reshape(-1), out, col_offset=2)
    ref = table.index_select(0, ids.reshape(-1)).reshape(6, 12)
    assert torch.allclose(out[:, 2:], ref)
    # scalar wide table
    wide = torch.randn(20, 1)
    s = ops.emb_gather_sum(wide, ids)
    ref_s = wide.reshape(-1).index_select(0, ids.reshape(-1)) \
        .reshape(6, 3).sum(dim=1)
    assert torch.allclose(s, ref_s, atol=1e-6)
    wide2 = wide.clone()
    g = torch.randn(6)
    ops.emb_scatter_sum(wide2, ids, g, alpha=-0.1)
    ref_w = wide.reshape(-1).clone()
    for b in range(6):
        for f in range(3):
            ref_w[ids[b, f]] += -0.1 * g[b]
    assert torch.allclose(wide2.reshape(-1), ref_w, atol=1e-5)


def test_emb_bwd_dense_cpu():
    torch.manual_seed(33)
    gt = torch.zeros(10, 4)
    ids = torch.tensor([1, 1, 3])
    grad = torch.randn(3, 4)
    ops.emb_bwd_dense(gt, ids, grad, scale=0.5)
    ref = torch.zeros(10, 4)
    ref.index_add_(0, ids, grad * 0.5)
    assert torch.allclose(gt, ref, atol=1e-6)
