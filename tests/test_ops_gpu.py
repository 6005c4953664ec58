"""GPU numerics: every HIP kernel vs a plain PyTorch fp32 reference.

All tests are @pytest.mark.gpu and run on a real MI355X via gpurun.
The HIP extension must be loaded — ops fail loudly on GPU without it.
"""

import pytest
import torch
from torch import nn

from tf_yarn_amd import ops

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")


@requires_gpu
def test_extension_is_native():
    """The native extension must actually be loaded on a GPU box."""
    assert ops.HAVE_EXT, "HIP extension not built/loaded on GPU box"
    import tf_yarn_amd.ops._C as C
    assert "_C" in C.__file__


@requires_gpu
@pytest.mark.parametrize("n", [1, 5, 1024, 1 << 20, (1 << 20) + 3])
def test_fused_sgd_gpu_vs_cpu_reference(n):
    torch.manual_seed(0)
    p_cpu = torch.randn(n)
    g_cpu = torch.randn(n)
    m_cpu = torch.randn(n).abs()
    p_gpu, g_gpu, m_gpu = (t.cuda() for t in (p_cpu, g_cpu, m_cpu))
    kwargs = dict(lr=0.1, momentum=0.9, dampening=0.0, weight_decay=0.01,
                  nesterov=True, first_step=False, grad_scale=0.5)
    ops.fused_sgd(p_cpu, g_cpu, m_cpu, None, **kwargs)
    ops.fused_sgd(p_gpu, g_gpu, m_gpu, None, **kwargs)
    assert torch.allclose(p_gpu.cpu(), p_cpu, atol=1e-6)
    assert torch.allclose(m_gpu.cpu(), m_cpu, atol=1e-6)


@requires_gpu
def test_fused_sgd_bf16_grad_and_copy():
    torch.manual_seed(1)
    n = 4096 + 1
    master = torch.randn(n).cuda()
    grad = torch.randn(n).cuda().to(torch.bfloat16)
    bf16_copy = master.to(torch.bfloat16)
    ref = master - 0.1 * grad.float()
    ops.fused_sgd(master, grad, None, bf16_copy, lr=0.1)
    assert torch.allclose(master, ref, atol=1e-6)
    assert torch.allclose(bf16_copy.float(),
                          ref.to(torch.bfloat16).float())


@requires_gpu
def test_fused_adam_gpu_vs_torch():
    torch.manual_seed(2)
    n = 100_003
    p0 = torch.randn(n)
    g0 = torch.randn(n)
    # torch reference on CPU fp32
    p_ref = p0.clone().requires_grad_(True)
    opt = torch.optim.Adam([p_ref], lr=0.01, weight_decay=0.001)
    p_ref.grad = g0.clone()
    opt.step()
    # ours on GPU
    p = p0.cuda()
    m = torch.zeros(n).cuda()
    v = torch.zeros(n).cuda()
    ops.fused_adam(p, g0.cuda(), m, v, None, lr=0.01,
                   weight_decay=0.001, step=1)
    assert torch.allclose(p.cpu(), p_ref.detach(), atol=1e-5)


@requires_gpu
def test_fused_adagrad_adadelta_gpu_vs_torch():
    torch.manual_seed(3)
    n = 50_001
    for name in ("adagrad", "adadelta"):
        p0 = torch.randn(n)
        g0 = torch.randn(n)
        p_ref = p0.clone().requires_grad_(True)
        if name == "adagrad":
            topt = torch.optim.Adagrad([p_ref], lr=0.05)
        else:
            topt = torch.optim.Adadelta([p_ref], lr=0.9)
        for _ in range(3):
            p_ref.grad = g0.clone()
            topt.step()
        p = p0.cuda()
        if name == "adagrad":
            acc = torch.zeros(n).cuda()
            for _ in range(3):
                ops.fused_adagrad(p, g0.cuda(), acc, lr=0.05)
        else:
            sq = torch.zeros(n).cuda()
            ad = torch.zeros(n).cuda()
            for _ in range(3):
                ops.fused_adadelta(p, g0.cuda(), sq, ad, lr=0.9)
        assert torch.allclose(p.cpu(), p_ref.detach(), atol=1e-5), name


@requires_gpu
@pytest.mark.parametrize("dim", [16, 4, 1, 7])
def test_emb_fwd_gpu(dim):
    torch.manual_seed(4)
    table = torch.randn(500, dim).cuda()
    ids = torch.randint(0, 500, (1000,)).cuda()
    out = ops.emb_fwd(table, ids)
    ref = table.index_select(0, ids)
    assert torch.equal(out, ref)
    out16 = ops.emb_fwd(table, ids, out_bf16=True)
    assert out16.dtype == torch.bfloat16
    assert torch.allclose(out16.float(),
                          ref.to(torch.bfloat16).float())


@requires_gpu
@pytest.mark.parametrize("dim,gdtype", [(16, torch.float32),
                                        (16, torch.bfloat16),
                                        (1, torch.float32)])
def test_emb_bwd_sgd_gpu(dim, gdtype):
    torch.manual_seed(5)
    table = torch.randn(200, dim).cuda()
    ref = table.clone()
    ids = torch.randint(0, 200, (4096,)).cuda()  # heavy collisions
    grad = torch.randn(4096, dim).cuda().to(gdtype)
    ops.emb_bwd_sgd(table, ids, grad, lr=0.1, scale=0.5)
    ref.index_add_(0, ids, grad.float(), alpha=-0.05)
    # atomics reorder fp32 adds: tolerance not equality
    assert torch.allclose(table, ref, atol=1e-3)


@requires_gpu
def test_bias_relu_gpu_fwd_bwd():
    torch.manual_seed(6)
    for dtype in (torch.float32, torch.bfloat16):
        x = torch.randn(64, 129).cuda().to(dtype)
        b = torch.randn(129).cuda().to(dtype)
        y = ops.bias_relu_fwd(x, b)
        ref = torch.relu((x.float() + b.float()))
        assert torch.allclose(y.float(), ref.to(dtype).float(), atol=1e-2)
        dy = torch.randn_like(y)
        dx = ops.bias_relu_bwd(dy, y)
        ref_dx = dy.float() * (y.float() > 0)
        assert torch.allclose(dx.float(), ref_dx.to(dtype).float(),
                              atol=1e-2)


@requires_gpu
def test_convert_scaled_gpu():
    src = torch.randn(100_001).cuda()
    dst = torch.empty(100_001, dtype=torch.bfloat16).cuda()
    ops.convert_scaled(src, dst, 0.125)
    assert torch.allclose(dst.float(),
                          (src * 0.125).to(torch.bfloat16).float())
    back = torch.empty(100_001).cuda()
    ops.convert_scaled(dst, back, 8.0)
    assert torch.allclose(back, dst.float() * 8.0)


@requires_gpu
def test_wide_deep_step_gpu_bf16():
    """Whole-model step on GPU with bf16 compute: finite loss, params move,
    loss decreases over a few steps on a fixed batch."""
    from tf_yarn_amd.models.synthetic import synthetic_criteo_batch
    from tf_yarn_amd.models.wide_deep import WideAndDeep
    from tf_yarn_amd.ops.optim import FusedSGD

    torch.manual_seed(7)
    tables = [1000] * 26
    model = WideAndDeep(table_sizes=tables, embedding_dim=16,
                        hidden=(128, 64),
                        compute_dtype=torch.bfloat16).cuda()
    opt = FusedSGD([p for p in model.parameters()
                    if not getattr(p, "_miyarn_sparse", False)], lr=0.05)
    dense, ids, labels = synthetic_criteo_batch(512, tables, device="cuda",
                                                seed=0)
    losses = []
    for _ in range(8):
        opt.zero_grad(set_to_none=False)
        logits = model(dense, ids)
        loss = nn.functional.binary_cross_entropy_with_logits(
            logits.float(), labels)
        loss.backward()
        opt.step()
        model.apply_sparse_updates(lr=0.05)
        losses.append(loss.item())
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0], f"loss did not decrease: {losses}"


@requires_gpu
def test_fused_sgd_mt_matches_per_tensor():
    torch.manual_seed(8)
    sizes = [437, 1024 * 100, 3, 64 * 64]
    ps1 = [torch.randn(n).cuda() for n in sizes]
    ps2 = [p.clone() for p in ps1]
    gs = [torch.randn(n).cuda().to(torch.bfloat16) for n in sizes]
    ms1 = [torch.randn(n).cuda().abs() for n in sizes]
    ms2 = [m.clone() for m in ms1]
    kwargs = dict(lr=0.1, momentum=0.9, weight_decay=0.01,
                  first_step=False)
    for p, g, m in zip(ps1, gs, ms1):
        ops.fused_sgd(p, g, m, None, **kwargs)
    ops.fused_sgd_mt(ps2, gs, ms2, None, **kwargs)
    for p1, p2 in zip(ps1, ps2):
        assert torch.equal(p1, p2)
    for m1, m2 in zip(ms1, ms2):
        assert torch.equal(m1, m2)


@requires_gpu
def test_fused_sgd_mt_many_tensors_chunking():
    torch.manual_seed(9)
    n_t = 30  # crosses the MT_MAX=24 per-launch boundary
    ps = [torch.randn(100 + i).cuda() for i in range(n_t)]
    refs = [p.clone() for p in ps]
    gs = [torch.randn_like(p) for p in ps]
    ops.fused_sgd_mt(ps, gs, None, None, lr=0.5)
    for p, r, g in zip(ps, refs, gs):
        assert torch.allclose(p, r - 0.5 * g, atol=1e-6)


@requires_gpu
def test_bias_relu_bwd_db_matches_reference():
    import tf_yarn_amd.ops._C as C
    torch.manual_seed(10)
    # cols=516 exercises the quad kernel, 512/1024 the bf16 oct kernel
    for dtype, cols in [(torch.float32, 516), (torch.bfloat16, 516),
                        (torch.bfloat16, 512), (torch.bfloat16, 1024)]:
        y = torch.relu(torch.randn(1024, cols)).cuda().to(dtype)
        dy = torch.randn(1024, cols).cuda().to(dtype)
        dx, dbias = C.bias_relu_bwd_db(dy.contiguous(), y.contiguous())
        ref_dx = dy.float() * (y.float() > 0)
        assert torch.allclose(dx.float(), ref_dx.to(dtype).float(),
                              atol=1e-2)
        ref_db = ref_dx.to(dtype).float().sum(dim=0)
        assert torch.allclose(dbias, ref_db, atol=0.5, rtol=1e-2)


@requires_gpu
def test_emb_gather_scatter_sum_gpu():
    torch.manual_seed(11)
    table = torch.randn(500, 1).cuda()
    ids = torch.randint(0, 500, (128, 26)).cuda()
    out = ops.emb_gather_sum(table, ids)
    ref = table.reshape(-1).index_select(
        0, ids.reshape(-1)).reshape(128, 26).sum(dim=1)
    assert torch.allclose(out, ref, atol=1e-4)
    out16 = ops.emb_gather_sum(table, ids, out_bf16=True)
    assert out16.dtype == torch.bfloat16

    grad = torch.randn(128).cuda()
    table2 = table.clone()
    ops.emb_scatter_sum(table2, ids, grad, alpha=-0.1)
    ref2 = table.clone()
    expanded = grad.reshape(-1, 1).expand(-1, 26).reshape(-1)
    ref2.reshape(-1).index_add_(0, ids.reshape(-1), expanded, alpha=-0.1)
    assert torch.allclose(table2, ref2, atol=1e-3)


@requires_gpu
def test_run_on_yarn_gpu_single():
    """End-to-end spawner run on the GPU box: chief task pinned to GPU 0,
    RCCL process group (world_size 1), HIP kernels in the training step."""
    import os
    import sys
    import tempfile

    import cloudpickle

    cloudpickle.register_pickle_by_value(sys.modules[__name__])
    from tf_yarn_amd import NodeLabel, TaskSpec
    from tf_yarn_amd.pytorch import run_on_yarn

    model_dir = tempfile.mkdtemp(prefix="miyarn_gpu_e2e_")

    def experiment_fn():
        import torch
        from torch import nn

        from tf_yarn_amd.models.synthetic import SyntheticCriteoDataset
        from tf_yarn_amd.models.wide_deep import WideAndDeep
        from tf_yarn_amd.ops.optim import FusedSGD
        from tf_yarn_amd.pytorch import DataLoaderArgs, PytorchExperiment

        tables = [1000] * 26

        def main_fn(model, loader, device, rank, tb_writer):
            module = model.module if hasattr(model, "module") else model
            assert device.startswith("cuda"), f"expected GPU, got {device}"
            opt = FusedSGD([p for p in module.parameters()
                            if not getattr(p, "_miyarn_sparse", False)],
                           lr=0.05)
            for dense, ids, labels in loader:
                dense = dense[0].to(device)
                ids = ids[0].to(device)
                labels = labels.reshape(-1).to(device)
                opt.zero_grad(set_to_none=False)
                loss = nn.functional.binary_cross_entropy_with_logits(
                    model(dense, ids).float(), labels)
                loss.backward()
                opt.step()
                module.apply_sparse_updates(0.05)
            assert torch.isfinite(loss)

        torch.manual_seed(0)
        model = WideAndDeep(table_sizes=tables, embedding_dim=16,
                            hidden=(64, 32),
                            compute_dtype=torch.bfloat16)
        return PytorchExperiment(
            model=model, main_fn=main_fn,
            train_dataset=SyntheticCriteoDataset(8 * 64, tables,
                                                 batch_size=64),
            dataloader_args=DataLoaderArgs(batch_size=1,
                                           pin_memory=False))

    metrics = run_on_yarn(
        experiment_fn,
        {"chief": TaskSpec(memory=2048, vcores=1, label=NodeLabel.GPU)},
        base_dir=model_dir)
    assert metrics is not None
    assert metrics.total_training_duration is not None


@requires_gpu
def test_col_reduce_dot_gpu():
    torch.manual_seed(12)
    for dtype in (torch.float32, torch.bfloat16):
        x = torch.randn(10000, 256).cuda().to(dtype)
        dy = torch.randn(10000).cuda().to(dtype)
        out = ops.col_reduce_dot(x, dy)
        ref = (x.float() * dy.float().unsqueeze(1)).sum(dim=0)
        assert torch.allclose(out, ref, atol=0.5, rtol=1e-2), dtype


@requires_gpu
def test_emb_fwd_into_gpu():
    torch.manual_seed(13)
    table = torch.randn(300, 16).cuda()
    ids = torch.randint(0, 300, (64, 26)).cuda()
    out = torch.full((64, 16 + 26 * 16), -9.0,
                     dtype=torch.bfloat16).cuda()
    ops.emb_fwd_into(table, ids, out, 16)
    ref = table.index_select(0, ids.reshape(-1)).reshape(
        64, 26 * 16).to(torch.bfloat16)
    assert torch.equal(out[:, 16:], ref)
    assert (out[:, :16] == -9.0).all()  # untouched prefix


@requires_gpu
def test_bias_relu_fwd_vec_gpu():
    torch.manual_seed(14)
    for dtype in (torch.float32, torch.bfloat16):
        x = torch.randn(1000, 512).cuda().to(dtype)
        b = torch.randn(512).cuda().to(dtype)
        y = ops.bias_relu_fwd(x, b)
        ref = torch.relu(x.float() + b.float()).to(dtype)
        assert torch.allclose(y.float(), ref.float(), atol=1e-2)


@requires_gpu
@pytest.mark.parametrize("B,N,M", [(4096, 128, 64), (8192, 1024, 448),
                                   (65536, 256, 512)])
def test_wgrad_nt_matches_reference(B, N, M):
    import tf_yarn_amd.ops._C as C
    torch.manual_seed(15)
    dy = (torch.randn(B, N, device="cuda") / 8).to(torch.bfloat16)
    x = (torch.randn(B, M, device="cuda") / 8).to(torch.bfloat16)
    out = C.wgrad_nt(dy, x, 0)
    ref = dy.float().t().mm(x.float())
    # bf16 inputs, fp32 accumulate both sides; tolerance scales with sqrt(K)
    assert out.shape == (N, M)
    err = (out - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err < 0.02 * max(1.0, scale), f"max err {err} scale {scale}"


@requires_gpu
def test_wgrad_nt_splitk_variants_agree():
    import tf_yarn_amd.ops._C as C
    torch.manual_seed(16)
    dy = (torch.randn(8192, 64, device="cuda") / 8).to(torch.bfloat16)
    x = (torch.randn(8192, 64, device="cuda") / 8).to(torch.bfloat16)
    a = C.wgrad_nt(dy, x, 1)
    b = C.wgrad_nt(dy, x, 8)
    assert torch.allclose(a, b, atol=1e-2, rtol=1e-3)


@requires_gpu
@pytest.mark.parametrize("B,N,M", [(4096, 128, 64), (8192, 1024, 432),
                                   (65536, 256, 512), (4096, 128, 136)])
def test_wgrad_nt128_matches_reference(B, N, M):
    import tf_yarn_amd.ops._C as C
    torch.manual_seed(17)
    dy = (torch.randn(B, N, device="cuda") / 8).to(torch.bfloat16)
    x = (torch.randn(B, M, device="cuda") / 8).to(torch.bfloat16)
    out = C.wgrad_nt128(dy, x, 0)
    ref = dy.float().t().mm(x.float())
    assert out.shape == (N, M)
    err = (out - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err < 0.02 * max(1.0, scale), f"max err {err} scale {scale}"


@requires_gpu
def test_estimator_dnn_classifier_gpu():
    """Estimator flavor on GPU: DNNClassifier + fused Adagrad, full
    train/evaluate/checkpoint cycle on cuda."""
    import tempfile

    import torch as t

    from tf_yarn_amd.estimator import DNNClassifier, RunConfig
    from tf_yarn_amd.ops.optim import FusedAdagrad

    model_dir = tempfile.mkdtemp(prefix="miyarn_est_gpu_")
    est = DNNClassifier(
        [32, 16], n_features=8, n_classes=2,
        optimizer_fn=lambda p: FusedAdagrad(p, lr=0.05),
        model_dir=model_dir, config=RunConfig(save_checkpoints_steps=20),
        device="cuda")

    def input_fn():
        t.manual_seed(0)
        for _ in range(30):
            x = t.randn(64, 8)
            yield x, (x.sum(dim=1) > 0).long()

    est.train(input_fn, max_steps=60)
    result = est.evaluate(input_fn, steps=10)
    assert result["accuracy"] > 0.8, result
    assert est.latest_checkpoint().endswith("model.ckpt-60")


@requires_gpu
def test_keras_model_gpu_fused_adadelta():
    """Keras shim on GPU with the fused Adadelta optimizer (the README's
    Keras optimizer)."""
    import torch as t
    from torch import nn

    from tf_yarn_amd.estimator.keras import KerasModel

    t.manual_seed(0)
    model = KerasModel(nn.Sequential(nn.Linear(8, 32), nn.ReLU(),
                                     nn.Linear(32, 1))).to("cuda")
    model.compile(optimizer="adadelta", loss="mse")
    x = t.randn(256, 8)
    y = x.sum(dim=1, keepdim=True)
    hist = model.fit(x, y, epochs=5, batch_size=32)
    assert hist["loss"][-1] < hist["loss"][0]


@requires_gpu
@pytest.mark.parametrize("B,N,M", [(4096, 256, 128), (8192, 1024, 432),
                                   (65536, 256, 512), (4096, 256, 264)])
def test_wgrad_nt256_matches_reference(B, N, M):
    import tf_yarn_amd.ops._C as C
    torch.manual_seed(18)
    dy = (torch.randn(B, N, device="cuda") / 8).to(torch.bfloat16)
    x = (torch.randn(B, M, device="cuda") / 8).to(torch.bfloat16)
    out = C.wgrad_nt256(dy, x, 0)
    ref = dy.float().t().mm(x.float())
    assert out.shape == (N, M)
    err = (out - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err < 0.02 * max(1.0, scale), f"max err {err} scale {scale}"


@requires_gpu
def test_emb_bwd_sgd_sorted_heavy_collisions():
    """The atomic-free sorted path must exactly sum duplicate-id grads
    (run-head segmented reduction, no atomics)."""
    import tf_yarn_amd.ops._C as C
    torch.manual_seed(19)
    table = torch.randn(50, 16).cuda()  # 50 rows, 65536 updates: runs ~1300
    ref = table.clone()
    ids = torch.randint(0, 50, (65536,)).cuda()
    grad = torch.randn(65536, 16).cuda()
    sorted_ids, perm = torch.sort(ids)
    g_sorted = grad.index_select(0, perm).contiguous()
    C.emb_bwd_sgd_sorted(table, sorted_ids, g_sorted, 0.1, 0.5)
    ref.index_add_(0, ids, grad, alpha=-0.05)
    assert torch.allclose(table, ref, atol=1e-2, rtol=1e-3)


@requires_gpu
def test_emb_bwd_sgd_wrapper_sorted_path_matches_atomic():
    import tf_yarn_amd.ops._C as C
    torch.manual_seed(20)
    table_a = torch.randn(1000, 16).cuda()
    table_b = table_a.clone()
    ids = torch.randint(0, 1000, (8192,)).cuda()
    grad = torch.randn(8192, 16).cuda().to(torch.bfloat16)
    ops.emb_bwd_sgd(table_a, ids, grad, lr=0.2, scale=1.0)  # wrapper
    C.emb_bwd_sgd(table_b, ids, grad, 0.2, 1.0)             # atomic path
    assert torch.allclose(table_a, table_b, atol=1e-3)


@requires_gpu
@pytest.mark.parametrize("M,N,K,relu", [
    (4096, 1024, 432, True),    # L1 forward (ragged K tail: 432 % 64 != 0)
    (4096, 512, 1024, True),    # L2 forward
    (4096, 256, 512, False),    # L3 shape, no epilogue
    (4096, 432, 1024, False),   # L1 dgrad (ragged N: B rows masked)
])
def test_gemm_bt_matches_reference(M, N, K, relu):
    import tf_yarn_amd.ops._C as C
    torch.manual_seed(21)
    a = (torch.randn(M, K, device="cuda") / 8).to(torch.bfloat16)
    b = (torch.randn(N, K, device="cuda") / 8).to(torch.bfloat16)
    bias = torch.randn(N, device="cuda").to(torch.bfloat16) if relu else None
    out = C.gemm_bt(a, b, bias, relu)
    ref = a.float() @ b.float().t()
    if bias is not None:
        ref = ref + bias.float()
    if relu:
        ref = torch.relu(ref)
    assert out.shape == (M, N) and out.dtype == torch.bfloat16
    err = (out.float() - ref).abs().max().item()
    scale = max(1.0, ref.abs().max().item())
    assert err < 0.03 * scale, f"max err {err} scale {scale}"


@requires_gpu
def test_linear_bias_relu_custom_fwd_gpu():
    """The fused gemm_bt forward path must agree with the lib path."""
    import os
    torch.manual_seed(22)
    x = (torch.randn(512, 432, device="cuda") / 4).to(
        torch.bfloat16).requires_grad_(True)
    w = (torch.randn(1024, 432, device="cuda") / 8).to(
        torch.bfloat16).requires_grad_(True)
    b = torch.randn(1024, device="cuda").to(torch.bfloat16).requires_grad_(True)
    os.environ["MIYARN_FWD"] = "custom"  # fused path is opt-in (slower)
    try:
        y = ops.linear_bias_relu(x, w, b)
        y.float().pow(2).sum().backward()
    finally:
        del os.environ["MIYARN_FWD"]
    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    y2 = ops.linear_bias_relu(x2, w2, b2)
    y2.float().pow(2).sum().backward()
    assert torch.allclose(y.float(), y2.float(), atol=0.05, rtol=0.05)
    assert torch.allclose(x.grad.float(), x2.grad.float(),
                          atol=0.1, rtol=0.1)
    assert torch.allclose(w.grad.float(), w2.grad.float(),
                          atol=0.5, rtol=0.1)


@requires_gpu
@pytest.mark.parametrize("rows,cols,dtype", [
    (4096, 256, torch.bfloat16), (1000, 64, torch.float32),
    (512, 512, torch.bfloat16), (333, 12, torch.bfloat16),
])
def test_row_dot_matches_reference(rows, cols, dtype):
    import tf_yarn_amd.ops._C as C
    torch.manual_seed(24)
    x = (torch.randn(rows, cols, device="cuda") / 4).to(dtype)
    w = (torch.randn(cols, device="cuda") / 4).to(dtype)
    bias = torch.randn(1, device="cuda").to(dtype)
    ref = x.float() @ w.float() + bias.float()
    out = ops.row_dot(x, w, bias)
    assert out.shape == (rows,) and out.dtype == dtype
    err = (out.float() - ref).abs().max().item()
    assert err < 0.02 * max(1.0, ref.abs().max().item())
    out2 = ops.row_dot(x, w, None)
    assert torch.allclose(out2.float(), (x.float() @ w.float()),
                          atol=0.02 * max(1.0, ref.abs().max().item()))


@requires_gpu
def test_scalar_head_ragged_width_padded_path():
    """ScalarHead(13) on GPU (padded row_dot path) must match the plain
    matmul reference, including grads."""
    from tf_yarn_amd.models.wide_deep import ScalarHead
    torch.manual_seed(25)
    head = ScalarHead(13, dtype=torch.bfloat16).cuda()
    x = (torch.randn(512, 13, device="cuda") / 4).to(torch.bfloat16)
    y = head(x)
    ref = x @ head.weight + head.bias
    assert torch.allclose(y.float(), ref.float(), atol=0.02)
    y.float().sum().backward()
    gw = head.weight.grad.clone()
    head.zero_grad()
    (x @ head.weight + head.bias).float().sum().backward()
    assert torch.allclose(gw.float(), head.weight.grad.float(),
                          atol=0.05, rtol=0.05)


@requires_gpu
def test_bce_head_fused_gpu():
    """Fused 3-part BCE head vs torch reference, forward and grads."""
    torch.manual_seed(26)
    n = 4096
    parts = [(torch.randn(n, device="cuda") / 4).to(torch.bfloat16)
             .requires_grad_(True) for _ in range(3)]
    labels = (torch.rand(n, device="cuda") > 0.5).float()
    loss = ops.bce_head_loss(*parts, labels)
    loss.backward()
    parts2 = [p.detach().clone().requires_grad_(True) for p in parts]
    z = (parts2[0].float() + parts2[1].float() + parts2[2].float())
    ref = torch.nn.functional.binary_cross_entropy_with_logits(z, labels)
    ref.backward()
    assert torch.allclose(loss, ref, atol=2e-3), (loss, ref)
    for p, p2 in zip(parts, parts2):
        assert torch.allclose(p.grad.float(), p2.grad.float(),
                              atol=1e-4), "grad mismatch"


# ---- binned scatter (round-2 kernel; spec: test_binned_scatter_spec) ------

@requires_gpu
@pytest.mark.parametrize("n_rows,n_upd", [
    (1_000_000, 200_000),   # bench-like sparsity (hash path)
    (4_000, 200_000),       # tiny table: every bin overflows (atomic path)
    (100, 64),              # single bin, small
])
@pytest.mark.parametrize("gdtype", [torch.float32, torch.bfloat16])
def test_emb_bwd_sgd_binned_matches_index_add(n_rows, n_upd, gdtype):
    torch.manual_seed(3)
    dim = 16
    table = torch.randn(n_rows, dim, device="cuda")
    ref = table.clone()
    ids = torch.randint(0, n_rows, (n_upd,), device="cuda")
    grad = torch.randn(n_upd, dim, device="cuda").to(gdtype)
    ops.emb_bwd_sgd_binned(table, ids, grad, lr=0.1, scale=0.5)
    ref.index_add_(0, ids, grad.float(), alpha=-0.05)
    atol = 1e-4 if gdtype == torch.float32 else 2e-2
    assert torch.allclose(table, ref, atol=atol), \
        (table - ref).abs().max().item()


@requires_gpu
def test_emb_bwd_sgd_binned_heavy_skew():
    """Power-law ids: one id takes ~25% of all updates (hash hot-slot +
    overflow fallback paths must still sum correctly)."""
    torch.manual_seed(4)
    n_rows, n_upd, dim = 100_000, 300_000, 16
    table = torch.randn(n_rows, dim, device="cuda")
    ref = table.clone()
    ids = torch.randint(0, n_rows, (n_upd,), device="cuda")
    ids[: n_upd // 4] = 7  # hot id
    ids[n_upd // 4: n_upd // 2] = torch.randint(
        0, 64, (n_upd // 4,), device="cuda")  # hot region
    grad = torch.randn(n_upd, dim, device="cuda")
    ops.emb_bwd_sgd_binned(table, ids, grad, lr=0.2, scale=1.0)
    ref.index_add_(0, ids, grad, alpha=-0.2)
    # hot row sums 75k grads: scale tolerance to the accumulated magnitude
    assert torch.allclose(table, ref, atol=1e-2), \
        (table - ref).abs().max().item()


@requires_gpu
@pytest.mark.parametrize("gdtype", [torch.float32, torch.bfloat16])
def test_emb_scatter_sum_binned_matches_reference(gdtype):
    torch.manual_seed(5)
    n_rows, batch, fan = 500_000, 20_000, 13
    table = torch.randn(n_rows, 1, device="cuda")
    ref = table.clone()
    ids = torch.randint(0, n_rows, (batch * fan,), device="cuda")
    gw = torch.randn(batch, device="cuda").to(gdtype)
    ops.emb_scatter_sum_binned(table, ids, gw, alpha=-0.05)
    expanded = gw.float().reshape(-1, 1).expand(-1, fan).reshape(-1)
    ref.reshape(-1).index_add_(0, ids, expanded, alpha=-0.05)
    atol = 1e-4 if gdtype == torch.float32 else 2e-2
    assert torch.allclose(table, ref, atol=atol), \
        (table - ref).abs().max().item()


@requires_gpu
def test_binned_permutation_matches_spec():
    """GPU pass A vs the executable spec (permutation + region grouping)."""
    torch.manual_seed(6)
    n_rows, n_upd, bits = 300_000, 50_000, 12
    ids = torch.randint(0, n_rows, (n_upd,), device="cuda")
    order, starts = ops.binned_permutation(ids, n_rows, bits)
    # order packs (row << 31 | update index)
    j = (order & ((1 << 31) - 1)).cpu()
    rows = (order >> 31).cpu()
    starts = starts.cpu().long()
    ids_cpu = ids.cpu()
    assert sorted(j.tolist()) == list(range(n_upd))
    assert (ids_cpu[j] == rows).all()  # packed rows match the ids
    for b in range(starts.numel() - 1):
        sl = j[starts[b]:starts[b + 1]]
        if sl.numel():
            assert ((ids_cpu[sl] >> bits) == b).all()


@requires_gpu
def test_sharded_embedding_binned_vs_atomic_path():
    """The module-level integration: _apply_pending with the binned path
    must produce the same tables as the atomic path."""
    import os
    from tf_yarn_amd.models.sharded_embedding import \
        ShardedCriteoEmbeddings
    torch.manual_seed(7)
    tables = [30_000] * 8
    dim, B = 16, 4096
    results = {}
    for flag in ("1", "0"):
        os.environ["MIYARN_BINNED_SCATTER"] = flag
        torch.manual_seed(7)
        emb = ShardedCriteoEmbeddings(tables, dim).cuda()
        ids = torch.randint(0, 30_000, (B, 8), device="cuda")
        buf = torch.zeros(B, 8 * dim, device="cuda")
        out, wide = emb(ids, buf, 0)
        (out.float().pow(2).mean() + wide.float().pow(2).mean()).backward()
        emb.apply_sparse_updates(0.3)
        results[flag] = (emb.weight.detach().clone(),
                         emb.wide_weight.detach().clone())
    os.environ.pop("MIYARN_BINNED_SCATTER", None)
    assert torch.allclose(results["1"][0], results["0"][0], atol=1e-4)
    assert torch.allclose(results["1"][1], results["0"][1], atol=1e-4)


@requires_gpu
@pytest.mark.parametrize("gdtype", [torch.float32, torch.bfloat16])
def test_emb_bwd_sgd_fused_wide_matches_separate(gdtype):
    torch.manual_seed(8)
    n_rows, batch, fan, dim = 200_000, 10_000, 13, 16
    table = torch.randn(n_rows, dim, device="cuda")
    wide = torch.randn(n_rows, 1, device="cuda")
    ref_t, ref_w = table.clone(), wide.clone()
    ids = torch.randint(0, n_rows, (batch * fan,), device="cuda")
    grad = torch.randn(batch * fan, dim, device="cuda").to(gdtype)
    gw = torch.randn(batch, device="cuda").to(gdtype)
    ops.emb_bwd_sgd_fused_wide(table, wide, ids, grad, gw,
                               lr=0.1, scale=0.5)
    ref_t.index_add_(0, ids, grad.float(), alpha=-0.05)
    expanded = gw.float().reshape(-1, 1).expand(-1, fan).reshape(-1)
    ref_w.reshape(-1).index_add_(0, ids, expanded, alpha=-0.05)
    atol = 1e-4 if gdtype == torch.float32 else 2e-2
    assert torch.allclose(table, ref_t, atol=atol)
    assert torch.allclose(wide, ref_w, atol=atol)


@requires_gpu
@pytest.mark.parametrize("out_bf16", [False, True])
def test_reduce_splitk_matches_torch_sum(out_bf16):
    import tf_yarn_amd.ops._C as C
    torch.manual_seed(9)
    part = torch.randn(28, 1024, 432, device="cuda")
    got = C.reduce_splitk(part, out_bf16)
    want = part.sum(0)
    assert got.dtype == (torch.bfloat16 if out_bf16 else torch.float32)
    atol = 5e-2 if out_bf16 else 1e-3
    assert torch.allclose(got.float(), want, atol=atol, rtol=1e-2)


@requires_gpu
def test_wgrad_out_bf16_matches_fp32_cast():
    import tf_yarn_amd.ops._C as C
    torch.manual_seed(10)
    dy = torch.randn(4096, 256, device="cuda").to(torch.bfloat16)
    x = torch.randn(4096, 512, device="cuda").to(torch.bfloat16)
    f32 = C.wgrad_nt256(dy, x, 8)
    bf = C.wgrad_nt256(dy, x, 8, True)
    assert bf.dtype == torch.bfloat16
    assert torch.allclose(bf.float(), f32.to(torch.bfloat16).float())


@requires_gpu
@pytest.mark.parametrize("relu,with_bias", [(True, True), (False, True),
                                            (False, False)])
def test_lt_linear_matches_reference(relu, with_bias):
    import tf_yarn_amd.ops._C as C
    torch.manual_seed(11)
    M, K, N = 4096, 432, 1024
    x = torch.randn(M, K, device="cuda").to(torch.bfloat16)
    w = (torch.randn(N, K, device="cuda") * 0.05).to(torch.bfloat16)
    b = torch.randn(N, device="cuda").to(torch.bfloat16) if with_bias \
        else None
    y = C.lt_linear(x, w, b, relu)
    ref = x.float() @ w.float().t()
    if with_bias:
        ref = ref + b.float()
    if relu:
        ref = torch.relu(ref)
    rel = ((y.float() - ref).abs().max()
           / ref.abs().max().clamp_min(1e-6)).item()
    assert rel < 5e-2, rel


@requires_gpu
def test_linear_bias_relu_lt_path_grads_match_kernel_path():
    """Forward via the fused hipBLASLt epilogue must produce the same
    gradients as the matmul+epilogue-kernel path."""
    import os
    M, K, N = 2048, 512, 256
    results = {}
    for flag in ("1", "0"):
        os.environ["MIYARN_LT_FWD"] = flag
        torch.manual_seed(12)
        x = torch.randn(M, K, device="cuda").to(torch.bfloat16)
        x.requires_grad_(True)
        torch.manual_seed(13)
        w = torch.nn.Parameter(
            (torch.randn(N, K, device="cuda") * 0.05).to(torch.bfloat16))
        b = torch.nn.Parameter(
            torch.randn(N, device="cuda").to(torch.bfloat16))
        y = ops.linear_bias_relu(x, w, b)
        y.float().pow(2).mean().backward()
        results[flag] = (y.detach().float(), x.grad.float(),
                         w.grad.float(), b.grad.float())
    os.environ.pop("MIYARN_LT_FWD", None)
    for a, c in zip(results["1"], results["0"]):
        assert torch.allclose(a, c, atol=5e-2, rtol=5e-2), \
            (a - c).abs().max().item()


@requires_gpu
def test_scatter_auto_skew_selection():
    """Unset MIYARN_BINNED_SCATTER = auto: uniform ids keep the atomic
    path, heavily duplicated ids flip to the binned-dedup path — probed
    once per module from the first update batch."""
    import os
    from tf_yarn_amd.models.sharded_embedding import \
        ShardedCriteoEmbeddings
    os.environ.pop("MIYARN_BINNED_SCATTER", None)

    def run_one(ids):
        torch.manual_seed(21)
        emb = ShardedCriteoEmbeddings([50_000] * 4, 16).cuda()
        buf = torch.zeros(ids.shape[0], 4 * 16, device="cuda")
        out, wide = emb(ids, buf, 0)
        (out.float().pow(2).mean() + wide.float().pow(2).mean()).backward()
        emb.apply_sparse_updates(0.1)
        return emb._binned_auto

    uniform = torch.randint(0, 50_000, (8192, 4), device="cuda")
    assert run_one(uniform) is False
    skewed = torch.randint(0, 64, (8192, 4), device="cuda")  # dup >> 2
    assert run_one(skewed) is True
