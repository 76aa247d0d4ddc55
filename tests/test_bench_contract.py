"""The driver contract: bench.py emits one valid JSON line (CPU mode)."""
import json
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_bench_cpu_smoke():
    out = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--cpu", "--smoke", "--no-rtt",
         "--size-mb", "0.25", "--keys-per-server", "4"],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    j = json.loads(lines[0])
    assert j["unit"] == "GB/s"
    assert j["value"] > 0
    assert j["n_gpus"] == 1
    assert j["higher_is_better"] is True
    assert j["data"] == "synthetic"
    # the metric string is the driver's join key against BASELINE.json
    baseline = json.loads((REPO / "BASELINE.json").read_text())
    assert j["metric"] == baseline["metric"], (j["metric"], baseline["metric"])


def test_rn50_buckets_shape():
    from ps_lite_amd.models import resnet50_grad_buckets, resnet50_param_sizes

    total = sum(resnet50_param_sizes())
    assert abs(total - 25_557_032) < 60_000, total  # ~25.5M params
    buckets = resnet50_grad_buckets()
    assert sum(buckets) == total * 4
    assert max(buckets) <= 4 << 20


def test_bench_cpu_rn50_contract():
    """`bench.py --cpu --mode rn50` runs BytePS reduce rounds on the CPU
    reduce handle and emits the JSON contract line."""
    out = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--cpu", "--mode", "rn50",
         "--smoke", "--no-rtt"],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    j = json.loads(lines[0])
    assert j["config"]["mode"] == "rn50-cpu"
    assert j["value"] > 0
