"""The examples/ scripts run end-to-end at smoke scale on CPU."""
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args, cwd):
    r = subprocess.run([sys.executable] + args, cwd=cwd,
                       capture_output=True, text=True, timeout=420)
    assert r.returncode == 0, r.stdout[-800:] + r.stderr[-800:]
    return r.stdout


@pytest.mark.timeout(500)
def test_uncond_then_sample(tmp_path):
    out = _run([os.path.join(ROOT, "examples/train_uncond_64px.py"),
                "--tiny", "--steps", "2"], cwd=str(tmp_path))
    assert "samples:" in out
    out = _run([os.path.join(ROOT, "examples/sample_from_checkpoint.py"),
                str(tmp_path / "checkpoints" / "uncond64"),
                "--num", "2", "--resolution", "16", "--steps", "2"],
               cwd=str(tmp_path))
    assert "samples: (2, 16, 16, 3)" in out


@pytest.mark.timeout(500)
def test_text_conditional(tmp_path):
    out = _run([os.path.join(ROOT, "examples/train_text_conditional.py"),
                "--steps", "2"], cwd=str(tmp_path))
    assert "CFG samples:" in out


@pytest.mark.timeout(500)
def test_latent_diffusion(tmp_path):
    out = _run([os.path.join(ROOT, "examples/latent_diffusion.py"),
                "--steps", "2"], cwd=str(tmp_path))
    assert "ldm trained" in out
