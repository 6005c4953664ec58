"""Wide-and-deep model tests on CPU (GPU numerics in test_ops_gpu.py)."""

import torch
from torch import nn

from tf_yarn_amd.models.synthetic import (SyntheticCriteoDataset,
                                          synthetic_criteo_batch)
from tf_yarn_amd.models.wide_deep import SparseEmbedding, WideAndDeep

TABLES = [100] * 26


def test_forward_shapes():
    model = WideAndDeep(table_sizes=TABLES, embedding_dim=8,
                        hidden=(32, 16))
    dense, ids, labels = synthetic_criteo_batch(4, TABLES, seed=0)
    logits = model(dense, ids)
    assert logits.shape == (4,)


def test_backward_and_sparse_updates():
    model = WideAndDeep(table_sizes=TABLES, embedding_dim=8,
                        hidden=(32, 16))
    dense, ids, labels = synthetic_criteo_batch(16, TABLES, seed=1)
    logits = model(dense, ids)
    loss = nn.functional.binary_cross_entropy_with_logits(
        logits.float(), labels)
    loss.backward()
    # dense params have grads; sparse tables have pending sinks instead
    assert model.head.weight.grad is not None
    assert model.deep_embedding.weight.grad is None
    assert len(model.deep_embedding.pending_grads()) == 1
    before = model.deep_embedding.weight.detach().clone()
    model.apply_sparse_updates(lr=0.1)
    after = model.deep_embedding.weight.detach()
    assert not torch.equal(before, after)
    assert len(model.deep_embedding.pending_grads()) == 0


def test_sparse_update_equals_autograd_reference():
    """The stashed-grad + fused scatter path must equal a plain autograd
    nn.Embedding SGD step on the same data."""
    torch.manual_seed(3)
    emb = SparseEmbedding([50, 60], dim=8)
    ref_weight = emb.weight.detach().clone().requires_grad_(True)
    ids = torch.tensor([[3, 7], [3, 12], [49, 0]])
    # our path
    out = emb(ids)
    out.pow(2).sum().backward()
    emb.apply_sparse_updates(lr=0.25)
    # reference: same math via autograd on a dense table
    flat = (ids + emb.offsets.unsqueeze(0)).reshape(-1)
    ref_out = ref_weight.index_select(0, flat).reshape(3, 16)
    ref_out.pow(2).sum().backward()
    expected = ref_weight.detach() - 0.25 * ref_weight.grad
    assert torch.allclose(emb.weight.detach(), expected, atol=1e-6)


def test_synthetic_dataset_deterministic():
    ds = SyntheticCriteoDataset(64, TABLES, batch_size=8)
    assert len(ds) == 8
    a = ds[3]
    b = ds[3]
    assert torch.equal(a[0], b[0]) and torch.equal(a[1], b[1])
    ids = a[1]
    assert ids.max() < 100 and ids.min() >= 0


def test_bf16_compute_mode_cpu_fallback():
    """bf16 mode still runs on CPU (falls back to fp32-ish torch ops)."""
    model = WideAndDeep(table_sizes=TABLES, embedding_dim=8,
                        hidden=(16,), compute_dtype=torch.bfloat16)
    dense, ids, labels = synthetic_criteo_batch(4, TABLES, seed=2)
    logits = model(dense, ids)
    assert logits.dtype == torch.bfloat16
    loss = nn.functional.binary_cross_entropy_with_logits(
        logits.float(), labels)
    loss.backward()
    model.apply_sparse_updates(lr=0.1)


def test_resnet_forward_backward_cpu():
    from tf_yarn_amd.models.resnet import resnet18_like, resnet50
    import torch
    from torch import nn
    m = resnet18_like(num_classes=10)
    x = torch.randn(2, 3, 64, 64)
    out = m(x)
    assert out.shape == (2, 10)
    nn.functional.cross_entropy(out, torch.tensor([1, 2])).backward()
    # resnet50 has the canonical parameter count (~25.6M)
    n_params = sum(p.numel() for p in resnet50().parameters())
    assert abs(n_params - 25_557_032) < 1000, n_params


def test_forward_with_labels_matches_separate_loss():
    """model(dense, ids, labels=...) (fused BCE head) must equal
    BCEWithLogits over the plain logits path."""
    from tf_yarn_amd.models.synthetic import synthetic_criteo_batch
    from tf_yarn_amd.models.wide_deep import WideAndDeep

    torch.manual_seed(11)
    tables = [50] * 26
    for sharded in (False, True):
        model = WideAndDeep(table_sizes=tables, embedding_dim=4,
                            hidden=(16, 8), sharded=sharded)
        dense, ids, labels = synthetic_criteo_batch(32, tables, seed=3)
        logits = model(dense, ids)
        ref = torch.nn.functional.binary_cross_entropy_with_logits(
            logits.float(), labels)
        model.clear_pending()
        fused = model(dense, ids, labels=labels)
        model.clear_pending()
        assert torch.allclose(fused, ref, atol=1e-5), (sharded, fused, ref)
        fused.backward()  # grads flow through all three parts
        assert model.wide_dense.weight.grad is not None
        assert model.head.weight.grad is not None


def test_keras_fit_trains_sink_based_embeddings():
    """The Keras fit loop must apply the sink-based sparse updates (CTR
    embedding grads never land in p.grad) and leave the sink drained."""
    import torch

    from tf_yarn_amd.estimator.keras import KerasModel
    from tf_yarn_amd.models.wide_deep import FlatInputWideAndDeep

    torch.manual_seed(0)
    model = KerasModel(FlatInputWideAndDeep(
        dense_dim=4, table_sizes=[50] * 3, embedding_dim=8,
        hidden=(16,)))
    model.compile(optimizer="sgd", loss="binary_crossentropy")
    net = model.module.net
    before = net.deep_embedding.weight.detach().clone()
    wide_before = net.wide_embedding.weight.detach().clone()
    g = torch.Generator().manual_seed(3)
    x = torch.cat([torch.randn(64, 4, generator=g),
                   torch.randint(0, 50, (64, 3), generator=g).float()],
                  dim=1)
    y = (torch.rand(64, generator=g) < 0.5).float()
    model.fit(x, y, epochs=1, batch_size=16)
    assert not torch.equal(net.deep_embedding.weight, before), \
        "deep embedding table did not train through Keras fit"
    assert not torch.equal(net.wide_embedding.weight, wide_before), \
        "wide table did not train through Keras fit"
    assert not net.deep_embedding.pending_grads(), "sink leaked"
    assert not net.wide_embedding.pending_grads(), "sink leaked"


def test_estimator_train_applies_sparse_updates(tmp_path):
    """Same contract through the Estimator train loop."""
    import torch

    from tf_yarn_amd.estimator import RunConfig
    from tf_yarn_amd.estimator.estimator import Estimator
    from tf_yarn_amd.models.wide_deep import FlatInputWideAndDeep

    torch.manual_seed(0)

    def module_fn():
        return FlatInputWideAndDeep(dense_dim=4, table_sizes=[50] * 3,
                                    embedding_dim=8, hidden=(16,))

    est = Estimator(
        module_fn=module_fn,
        loss_fn=lambda o, t: torch.nn.functional
        .binary_cross_entropy_with_logits(o.float(), t.float()),
        optimizer_fn=lambda ps: torch.optim.SGD(
            [p for p in ps if not getattr(p, "_miyarn_sparse", False)],
            lr=0.1),
        model_dir=str(tmp_path), config=RunConfig())

    def input_fn():
        g = torch.Generator().manual_seed(4)
        for _ in range(4):
            x = torch.cat(
                [torch.randn(32, 4, generator=g),
                 torch.randint(0, 50, (32, 3), generator=g).float()],
                dim=1)
            y = (torch.rand(32, generator=g) < 0.5).float()
            yield x, y

    est._ensure_built()
    net = est._module.net
    before = net.deep_embedding.weight.detach().clone()
    est.train(input_fn, steps=4, save_checkpoints=False)
    assert not torch.equal(net.deep_embedding.weight, before), \
        "embedding table did not train through Estimator.train"
    assert not net.deep_embedding.pending_grads(), "sink leaked"
