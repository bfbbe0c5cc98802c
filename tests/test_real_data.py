"""End-to-end training on REAL data (the reference's actual task).

The reference trains fetch_openml MNIST to high accuracy
(/root/reference/train.py:148-152, data via download_dataset.py:10-23).
This environment has no network, so the real-data stand-in is
sklearn's bundled load_digits (1,797 real handwritten 8×8 digits)
upsampled to the reference's 784-feature shape and written in the
reference's exact on-disk format by `prepare_data.py --source digits`.
The gate mirrors the reference's convergence expectation: high
validation accuracy after a short training run through the full CLI.
"""

import re
import subprocess
import sys

ROOT = __file__.rsplit("/tests/", 1)[0]


def _run(cmd):
    r = subprocess.run(cmd, cwd=ROOT, capture_output=True, text=True,
                       timeout=600)
    assert r.returncode == 0, r.stdout + r.stderr
    return r.stdout


def test_real_digits_convergence_cli(tmp_path):
    d = str(tmp_path / "digits")
    _run([sys.executable, "prepare_data.py", "--out", d, "--source", "digits"])
    out = _run([sys.executable, "train.py", "--data-dir", d,
                "--epochs", "40", "--loss", "xent", "--lr", "0.05",
                "--momentum", "0.9", "--schedule", "gpipe",
                "--mubatches", "4", "--device", "cpu"])
    accs = [float(m) for m in re.findall(r"val_acc=([0-9.]+)", out)]
    assert accs, out
    assert max(accs) >= 0.95, f"best val_acc {max(accs)} < 0.95\n{out[-2000:]}"


def test_lr_schedule_cli(tmp_path):
    """Warmup + cosine schedule runs end to end through the CLI."""
    out = _run([sys.executable, "train.py", "--epochs", "6", "--lr", "0.1",
                "--lr-schedule", "cosine", "--warmup-epochs", "2",
                "--loss", "xent", "--samples", "1024", "--device", "cpu"])
    accs = [float(m) for m in re.findall(r"val_acc=([0-9.]+)", out)]
    assert len(accs) >= 6 and accs[-1] > accs[0]
