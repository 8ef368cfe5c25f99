"""Guard the driver's bench.py contract: torchrun 2-rank CPU smoke must
print ONE valid JSON line with the required fields (the round-end harness
launches bench.py exactly this way on 1..8 GPUs)."""
import json
import os
import subprocess
import sys


def test_bench_torchrun_cpu_smoke():
    env = dict(os.environ)
    env["VESCALE_BENCH_BACKEND"] = "gloo"
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29655",
            "bench.py", "--model", "llama_tiny", "--steps", "2",
            "--warmup", "1", "--batch", "2", "--seq", "64",
        ],
        capture_output=True, text=True, timeout=420, env=env,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    for field in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                  "ms_per_step", "higher_is_better", "scaling",
                  "vs_baseline", "dtype", "data", "config"):
        assert field in d, field
    assert d["n_gpus"] == 2
    assert d["scaling"] == "weak"
    assert d["value"] > 0


def test_bench_torchrun_cpu_8rank_smoke():
    """8-rank gloo smoke of the EXACT launch shape the driver uses for the
    round-end SCALE run — tiny model, 1 step."""
    env = dict(os.environ)
    env["VESCALE_BENCH_BACKEND"] = "gloo"
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "8",
            "--master-addr", "127.0.0.1", "--master-port", "29656",
            "bench.py", "--model", "llama_tiny", "--steps", "1",
            "--warmup", "1", "--batch", "1", "--seq", "64",
        ],
        capture_output=True, text=True, timeout=600, env=env,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 8
    assert d["config"]["parallelism"] == "fsdp8"
    assert d["value"] > 0


def test_bench_torchrun_tp2_smoke():
    """The 2D TP x FSDP bench path (--tp) at ws=4 on gloo (BASELINE's
    'TP=2 x FSDP=4' config shape, scaled down)."""
    env = dict(os.environ)
    env["VESCALE_BENCH_BACKEND"] = "gloo"
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "4",
            "--master-addr", "127.0.0.1", "--master-port", "29657",
            "bench.py", "--model", "llama_tiny", "--steps", "1",
            "--warmup", "1", "--batch", "1", "--seq", "64", "--tp", "2",
        ],
        capture_output=True, text=True, timeout=600, env=env,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["config"]["parallelism"] == "tp2_fsdp2"
    assert d["value"] > 0
