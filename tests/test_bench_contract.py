"""The driver depends on bench.py's CLI + JSON contract — pin it."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--device", "cpu",
         "--steps", "2", "--warmup", "1", "--batch", "4"],
        capture_output=True, text=True, timeout=900, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, key
    assert d["n_gpus"] == 1
    assert d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["unit"] == "img/s"
    assert d["value"] > 0
    cfg = d["config"]
    assert cfg["model"] == "resnet18_cifar"
    assert cfg["grad_format"] == "e4m3"
    assert cfg["parallelism"] == "dp1"


def test_compat_namespace():
    out = subprocess.run(
        [sys.executable, "-c",
         "from CPDtorch.quant import float_quantize, quantizer, quant_gemm, "
         "Quantizer, Quant_Linear, Quant_Conv\n"
         "from CPDtorch.utils.dist_util import dist_init, DistModule, "
         "sum_gradients, normal_sum_gradients, kahan_sum_gradients\n"
         "import torch\n"
         "assert float_quantize(torch.tensor([255.9]), 4, 3).item() == 256.0\n"
         "print('ok')"],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "ok" in out.stdout


def test_bench_torchrun_w1_inits_group():
    """Launched with torchrun-style env at W=1, bench must init the process
    group (gloo on CPU; RCCL on GPU — hardware-validated at 30.0k img/s)
    and still emit the JSON contract line."""
    env = dict(os.environ, WORLD_SIZE="1", RANK="0", LOCAL_RANK="0",
               MASTER_ADDR="127.0.0.1", MASTER_PORT="29561")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--device", "cpu",
         "--steps", "2", "--warmup", "1", "--batch", "4"],
        capture_output=True, text=True, timeout=900, cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    d = json.loads(out.stdout.strip().splitlines()[-1])
    assert d["n_gpus"] == 1 and d["value"] > 0
