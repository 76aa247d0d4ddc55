"""Data-parallel gradient sync through the PS (CPU, 2 workers)."""
import numpy as np

from ps_lite_amd.parallel import launch_local


def _dp_worker(ps, rank):
    import torch

    from ps_lite_amd.parallel.dp import PSGradSync

    torch.manual_seed(42)  # same init on both workers
    model = torch.nn.Linear(16, 4)
    sync = PSGradSync(ps, ps.KVWorker(0, 0), model.parameters(), num_workers=2, device=-1)

    grads_seen = []
    for step in range(2):
        torch.manual_seed(100 * (rank + 1) + step)  # different data per worker
        x = torch.randn(8, 16)
        y = model(x).sum()
        model.zero_grad()
        y.backward()
        local = [p.grad.clone() for p in model.parameters()]
        sync.allreduce()
        grads_seen.append([p.grad.numpy().copy() for p in model.parameters()])
        # basic sanity: averaged grad differs from the local one
        assert not all(torch.allclose(l, p.grad) for l, p in zip(local, model.parameters()))
    return [[g.tolist() for g in step_g] for step_g in grads_seen]


def test_dp_allreduce_two_workers():
    results = launch_local(2, 2, _dp_worker, timeout=240)
    # both workers must end with identical averaged gradients each step
    for step in range(2):
        for g0, g1 in zip(results[0][step], results[1][step]):
            assert np.allclose(np.array(g0), np.array(g1), atol=1e-6)


def test_train_dp_example():
    """The examples/python/train_dp.py script must run and converge."""
    import os
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    p = subprocess.run([sys.executable, os.path.join(repo, "examples/python/train_dp.py")],
                       capture_output=True, text=True, timeout=280, cwd=repo)
    assert p.returncode == 0, (p.stdout[-1500:], p.stderr[-1500:])
    assert "converged" in p.stdout
