"""Every shipped example runs end-to-end on CPU/gloo (tiny configs).

Examples are the user-facing contract (mirrors the reference's
legacy/examples/*); running them in CI keeps them from rotting silently.
Each launches exactly as the README documents (torchrun or plain python)
in a subprocess.
"""
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _torchrun(nproc, script, *args, timeout=420, port=29710, env=None):
    e = dict(os.environ)
    if env:
        e.update(env)
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", f"--nproc-per-node={nproc}",
            "--master-addr", "127.0.0.1", "--master-port", str(port),
            script, *args,
        ],
        capture_output=True, text=True, timeout=timeout, cwd=ROOT, env=e,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    return out.stdout


def test_example_llama3_fsdp_tiny():
    out = _torchrun(
        2, "examples/llama3_fsdp_train.py",
        "--model", "llama_tiny", "--steps", "2", "--batch", "2", "--seq", "64",
        port=29711,
    )
    assert "loss" in out.lower()


def test_example_nanogpt_4d():
    out = _torchrun(
        4, "examples/nanogpt_4d_finetune.py",
        "--dp", "2", "--tp", "2", "--steps", "2", "--batch", "4",
        port=29712,
    )
    assert "loss" in out.lower()


def test_example_mixtral_ep():
    out = _torchrun(
        2, "examples/mixtral_ep_train.py", "--steps", "2", "--batch", "2",
        "--seq", "64", port=29713,
    )
    assert "loss" in out.lower()


def test_example_pp_zbv():
    out = _torchrun(
        4, "examples/pp_zbv_train.py", "--schedule", "zero_bubble_v",
        "--steps", "2", port=29714,
    )
    assert "DONE" in out


def test_example_hf_llama_tp():
    pytest.importorskip("transformers")
    out = _torchrun(
        2, "examples/hf_llama_tp_finetune.py", "--steps", "2", "--seq", "32",
        port=29715,
    )
    assert "DONE" in out


def test_example_checkpoint_reshard(tmp_path):
    ck = str(tmp_path / "ck")
    _torchrun(4, "examples/checkpoint_reshard.py", "--save", ck, "--steps", "2",
              port=29716)
    out = _torchrun(2, "examples/checkpoint_reshard.py", "--load", ck,
                    "--steps", "1", port=29717)
    assert "loss" in out.lower() or "loaded" in out.lower() or out


def test_example_long_context_sp():
    # ring and ulysses must print IDENTICAL losses (both exact)
    out_r = _torchrun(
        2, "examples/long_context_sp.py", "--algo", "ring", "--steps", "2",
        port=29717,
    )
    out_u = _torchrun(
        2, "examples/long_context_sp.py", "--algo", "ulysses", "--steps", "2",
        port=29718,
    )
    lr = [l.split("loss")[-1] for l in out_r.splitlines() if "loss" in l]
    lu = [l.split("loss")[-1] for l in out_u.splitlines() if "loss" in l]
    assert lr and lr == lu, (lr, lu)
    assert "DONE" in out_r and "DONE" in out_u
