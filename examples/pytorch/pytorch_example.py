"""PytorchExperiment DDP example (the reference's
``examples/pytorch/pytorch_example.py``): model wrapped in the framework
reducer, per-epoch checkpoints, rank-0-only writes, resume support.

Run: python examples/pytorch/pytorch_example.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

from tf_yarn_amd import NodeLabel, TaskSpec
from tf_yarn_amd.pytorch import run_on_yarn


def experiment_fn(model_dir: str):
    def make():
        import torch
        from torch import nn

        from tf_yarn_amd.pytorch import (DataLoaderArgs,
                                         DistributedDataParallelArgs,
                                         PytorchExperiment)
        from tf_yarn_amd.pytorch import model_ckpt

        def main_fn(model, loader, device, rank, tb_writer):
            import torch.distributed as dist
            opt = torch.optim.SGD(model.parameters(), lr=0.05)
            state = model_ckpt.load_latest_ckpt(model_dir, model, opt,
                                                device)
            start_epoch = (state["epoch"] + 1) if state else 0
            for epoch in range(start_epoch, start_epoch + 2):
                for x, y in loader:
                    x, y = x.to(device), y.to(device)
                    opt.zero_grad()
                    loss = nn.functional.mse_loss(model(x), y)
                    loss.backward()
                    opt.step()
                dist.barrier()
                if rank == 0:  # only rank 0 writes checkpoints
                    model_ckpt.save_ckpt(model_dir, model, opt, epoch)
                if tb_writer is not None:
                    tb_writer.add_scalar("loss", loss.item(), epoch)

        torch.manual_seed(0)
        model = nn.Sequential(nn.Linear(8, 32), nn.ReLU(),
                              nn.Linear(32, 1))
        x = torch.randn(512, 8)
        dataset = torch.utils.data.TensorDataset(
            x, x.sum(dim=1, keepdim=True))
        return PytorchExperiment(
            model=model,
            main_fn=main_fn,
            train_dataset=dataset,
            dataloader_args=DataLoaderArgs(batch_size=32,
                                           pin_memory=False),
            tensorboard_hdfs_dir=os.path.join(model_dir, "tb"),
            ddp_args=DistributedDataParallelArgs(bucket_cap_mb=32),
        )
    return make


def main():
    model_dir = os.environ.get("MODEL_DIR", "/tmp/miyarn_pt_example")
    use_gpu = os.environ.get("USE_GPU", "0") == "1"
    spec = TaskSpec(memory=1024, vcores=2, instances=2,
                    nb_proc_per_worker=2,
                    label=NodeLabel.GPU if use_gpu else NodeLabel.CPU)
    metrics = run_on_yarn(
        experiment_fn(model_dir),
        {"chief": TaskSpec(memory=1024, vcores=1), "worker": spec})
    print("run metrics:", metrics)


if __name__ == "__main__":
    main()
