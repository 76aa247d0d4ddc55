#!/usr/bin/env python3
"""Data-parallel training through the parameter server: 2 workers train
the same MLP on different minibatches, gradients averaged via
ZPush/ZPull each step (the BytePS pattern; CPU demo — on MI355X pass
device>=0 and the reduce-mode GPU handler takes over).

Run:  python examples/python/train_dp.py
"""
import sys

import numpy as np

sys.path.insert(0, __file__.rsplit("/examples/", 1)[0])
from ps_lite_amd.parallel import launch_local  # noqa: E402


def worker(ps, rank):
    import torch

    from ps_lite_amd.parallel.dp import PSGradSync

    torch.manual_seed(7)  # identical init on every worker
    model = torch.nn.Sequential(
        torch.nn.Linear(32, 64), torch.nn.ReLU(), torch.nn.Linear(64, 1))
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    sync = PSGradSync(ps, ps.KVWorker(0, 0), model.parameters(),
                      num_workers=2, device=-1)
    torch.manual_seed(1234)  # shared task: y = Xw + noise
    w_true = torch.randn(32, 1)
    losses = []
    for step in range(20):
        torch.manual_seed(1000 * (rank + 1) + step)  # different data shard
        x = torch.randn(64, 32)
        y = x @ w_true + 0.01 * torch.randn(64, 1)
        loss = torch.nn.functional.mse_loss(model(x), y)
        opt.zero_grad()
        loss.backward()
        sync.allreduce()  # grads now averaged across both workers
        opt.step()
        losses.append(float(loss))
    if rank == 0:
        print(f"loss: {losses[0]:.4f} -> {losses[-1]:.4f}")
    return losses


def main():
    results = launch_local(2, 2, worker, timeout=240)
    for rank in (0, 1):
        assert results[rank][-1] < results[rank][0] * 0.5, "did not converge"
    # identical averaged gradients + identical init => identical models:
    # both workers must report the same loss trajectory shape
    print("converged on both workers; final losses:",
          [round(results[r][-1], 4) for r in (0, 1)])


if __name__ == "__main__":
    main()
