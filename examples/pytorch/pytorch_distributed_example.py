"""Raw ``distributed`` flavor example (the reference's
``examples/pytorch/pytorch_distributed_example.py``): the library only
provides rank/master/world_size; the user function brings up the process
group, wraps the model and synchronizes with a barrier itself.

Run: python examples/pytorch/pytorch_distributed_example.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

from tf_yarn_amd import TaskSpec
from tf_yarn_amd.distributed import run_on_yarn


def train_fn(task_params):
    import os

    import torch
    import torch.distributed as dist
    from torch import nn

    from tf_yarn_amd.parallel.ddp import BucketedDataParallel

    os.environ["MASTER_ADDR"] = task_params.master_address
    os.environ["MASTER_PORT"] = str(task_params.master_port)
    # RCCL needs a distinct GPU per rank; on a box with fewer GPUs
    # than ranks, collectives stage through gloo (compute stays on GPU)
    backend = "nccl" if (task_params.gpu_id is not None
                         and torch.cuda.is_available()
                         and task_params.world_size
                         <= torch.cuda.device_count()) else "gloo"
    dist.init_process_group(backend, rank=task_params.rank,
                            world_size=task_params.world_size)
    device = (f"cuda:{task_params.gpu_id}"
              if task_params.gpu_id is not None
              and torch.cuda.is_available() else "cpu")
    torch.manual_seed(0)
    model = nn.Sequential(nn.Linear(16, 32), nn.ReLU(),
                          nn.Linear(32, 10)).to(device)
    ddp = BucketedDataParallel(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    for step in range(20):
        torch.manual_seed(100 + 10 * step + task_params.rank)
        x = torch.randn(32, 16, device=device)
        y = torch.randint(0, 10, (32,), device=device)
        opt.zero_grad()
        loss = nn.functional.cross_entropy(ddp(x), y)
        loss.backward()
        opt.step()
    dist.barrier()
    if task_params.rank == 0:
        print(f"final loss: {loss.item():.4f}")
    dist.destroy_process_group()


def main():
    metrics = run_on_yarn(
        train_fn,
        {
            "chief": TaskSpec(memory=1024, vcores=1),
            "worker": TaskSpec(memory=1024, vcores=1, instances=2),
        })
    print("run metrics:", metrics)


if __name__ == "__main__":
    main()
