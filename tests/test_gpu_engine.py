"""End-to-end entrypoint runs on the GPU (marked gpu; exercised by the
driver's round-end suite on a fresh MI355X)."""
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope="module", autouse=True)
def _need_gpu():
    assert torch.cuda.is_available()


def test_digits_entrypoint_gpu():
    r = subprocess.run(
        [sys.executable, "usps_mnist.py", "--synthetic", "--synthetic_size",
         "128", "--epochs", "1", "--group_size", "4", "--num_workers", "0",
         "--test_batch_size", "32", "--log_interval", "2", "--dtype",
         "bfloat16", "--loss", "mec"],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "Test set" in r.stdout


def test_officehome_entrypoint_gpu(tmp_path):
    ckpt = str(tmp_path / "oh.pt")
    r = subprocess.run(
        [sys.executable, "resnet50_dwt_mec_officehome.py", "--synthetic",
         "--synthetic_size", "48", "--num_iters", "3", "--source_batch_size",
         "8", "--test_batch_size", "16", "--num_workers", "0",
         "--check_acc_step", "100", "--log_interval", "1", "--stats_passes",
         "1", "--checkpoint_path", ckpt, "--checkpoint_every", "2"],
        cwd=REPO, capture_output=True, text=True, timeout=900)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "Accuracy" in r.stdout
    assert os.path.exists(ckpt)

    # resume path
    r2 = subprocess.run(
        [sys.executable, "resnet50_dwt_mec_officehome.py", "--synthetic",
         "--synthetic_size", "48", "--num_iters", "3", "--source_batch_size",
         "8", "--test_batch_size", "16", "--num_workers", "0",
         "--check_acc_step", "100", "--log_interval", "1", "--stats_passes",
         "1", "--checkpoint_path", ckpt, "--resume"],
        cwd=REPO, capture_output=True, text=True, timeout=900)
    assert r2.returncode == 0, r2.stderr[-2000:]
    assert "Resumed from" in r2.stdout


def test_officehome_hip_conv_path_gpu():
    """DWT_AMD_CONV=hip runs the backbone on the MFMA conv fwd+dgrad."""
    env = dict(os.environ, DWT_AMD_CONV="hip")
    r = subprocess.run(
        [sys.executable, "resnet50_dwt_mec_officehome.py", "--synthetic",
         "--synthetic_size", "24", "--num_iters", "2", "--source_batch_size",
         "8", "--test_batch_size", "8", "--num_workers", "0",
         "--check_acc_step", "100", "--log_interval", "1", "--stats_passes",
         "1"],
        cwd=REPO, capture_output=True, text=True, timeout=900, env=env)
    assert r.returncode == 0, r.stderr[-2000:]
