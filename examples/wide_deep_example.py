"""Flagship example: Criteo wide-and-deep CTR training across the node's
GPUs (BASELINE config 3 shape) - fused embedding gather/scatter HIP
kernels, bf16 compute, bucketed ring allreduce for the dense MLP, sparse
allgather sync for the embeddings.

Run (CPU plumbing): python examples/wide_deep_example.py
Run (GPU):          USE_GPU=1 N_GPUS=8 python examples/wide_deep_example.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tf_yarn_amd import NodeLabel, TaskSpec
from tf_yarn_amd.pytorch import run_on_yarn


def experiment_fn():
    def make():
        import torch
        from torch import nn

        from tf_yarn_amd.models import (SyntheticCriteoDataset, WideAndDeep)
        from tf_yarn_amd.ops.optim import FusedSGD
        from tf_yarn_amd.pytorch import DataLoaderArgs, PytorchExperiment

        use_gpu = torch.cuda.is_available()
        tables = [100_000] * 26
        lr = 0.02

        def main_fn(model, loader, device, rank, tb_writer):
            module = model.module if hasattr(model, "module") else model
            opt = FusedSGD(
                [p for p in module.parameters()
                 if not getattr(p, "_miyarn_sparse", False)], lr=lr)
            loss_fn = nn.BCEWithLogitsLoss()
            for step, (dense, ids, labels) in enumerate(loader):
                dense = dense[0].to(device) if dense.dim() == 3 \
                    else dense.to(device)
                ids = ids[0].to(device) if ids.dim() == 3 \
                    else ids.to(device)
                labels = labels.reshape(-1).to(device)
                opt.zero_grad(set_to_none=False)
                logits = model(dense, ids)
                loss = loss_fn(logits.float(), labels)
                loss.backward()
                opt.step()
                module.apply_sparse_updates(lr)
                if rank == 0 and step % 10 == 0:
                    print(f"step {step} loss {loss.item():.4f}")

        torch.manual_seed(0)
        model = WideAndDeep(
            table_sizes=tables, embedding_dim=16, hidden=(256, 128),
            compute_dtype=torch.bfloat16 if use_gpu else torch.float32)
        dataset = SyntheticCriteoDataset(
            64 * 64, tables, batch_size=64)
        return PytorchExperiment(
            model=model,
            main_fn=main_fn,
            train_dataset=dataset,
            dataloader_args=DataLoaderArgs(batch_size=1,
                                           pin_memory=False),
        )
    return make


def main():
    use_gpu = os.environ.get("USE_GPU", "0") == "1"
    n = int(os.environ.get("N_GPUS", "2"))
    label = NodeLabel.GPU if use_gpu else NodeLabel.CPU
    metrics = run_on_yarn(
        experiment_fn(),
        {
            "chief": TaskSpec(memory=2048, vcores=1, label=label),
            "worker": TaskSpec(memory=2048, vcores=1, instances=n - 1,
                               label=label),
        })
    print("run metrics:", metrics)


if __name__ == "__main__":
    main()
