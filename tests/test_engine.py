"""Engine smoke tests on CPU: losses decrease, eval runs, entrypoints wire up."""
import sys
import types

import pytest
import torch
from torch.utils.data import DataLoader

from dwt_amd.data import SyntheticDigits, SyntheticOfficeHome, Compose, Normalize, ToTensor
from dwt_amd.engine.digits import test as _eval_digits, train_digits_epoch
from dwt_amd.engine.officehome import (eval_pass_collect_stats,
                                       train_infinite_collect_stats)
from dwt_amd.models import LeNet, Bottleneck, ResNetDWT


def _args(**kw):
    ns = types.SimpleNamespace(log_interval=10, num_iters=3, check_acc_step=100)
    for k, v in kw.items():
        setattr(ns, k, v)
    return ns


def test_digits_training_loss_decreases():
    torch.manual_seed(0)
    t = Compose([ToTensor(), Normalize([0.5], [0.5])])
    src = SyntheticDigits(256, transform=t, seed=1)
    tgt = SyntheticDigits(256, transform=t, seed=2, shift=0.1)
    tst = SyntheticDigits(128, train=False, transform=t, seed=2, shift=0.1)
    src_loader = DataLoader(src, batch_size=32, shuffle=True, drop_last=True)
    tgt_loader = DataLoader(tgt, batch_size=32, shuffle=True, drop_last=True)
    test_loader = DataLoader(tst, batch_size=64)

    model = LeNet(group_size=4)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, weight_decay=5e-4)
    device = torch.device("cpu")
    args = _args()

    import torch.nn.functional as F

    def epoch_loss():
        model.train()
        tot, n = 0.0, 0
        with torch.no_grad():
            for (s, sy), (t_, _) in zip(src_loader, tgt_loader):
                out = model(torch.cat((s, t_), 0))
                so, _ = torch.split(out, out.shape[0] // 2, 0)
                tot += F.nll_loss(F.log_softmax(so, 1), sy).item()
                n += 1
        return tot / n

    before = epoch_loss()
    for ep in range(2):
        train_digits_epoch(args, model, device, src_loader, tgt_loader, opt, ep, 0.1)
    after = epoch_loss()
    assert after < before, (before, after)
    acc = _eval_digits(args, model, device, test_loader)
    assert acc > 15.0  # well above 10% chance after 2 epochs


def test_digits_mec_mode_runs():
    t = Compose([ToTensor(), Normalize([0.5], [0.5])])
    src = SyntheticDigits(64, transform=t, seed=1)
    tgt = SyntheticDigits(64, transform=t, transform_aug=t, seed=2)
    src_loader = DataLoader(src, batch_size=16, drop_last=True)
    tgt_loader = DataLoader(tgt, batch_size=16, drop_last=True)
    model = LeNet(group_size=4, streams=3)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    train_digits_epoch(_args(), model, torch.device("cpu"), src_loader,
                       tgt_loader, opt, 0, 0.1, loss_kind="mec")


@pytest.mark.slow
def test_officehome_loop_smoke(tmp_path):
    torch.manual_seed(0)
    src = SyntheticOfficeHome(12, num_classes=7, img_size=64, seed=1)
    tgt = SyntheticOfficeHome(12, num_classes=7, img_size=64, transform_aug=True, seed=2)
    tst = SyntheticOfficeHome(8, num_classes=7, img_size=64, seed=2)
    src_loader = DataLoader(src, batch_size=4, drop_last=True)
    tgt_loader = DataLoader(tgt, batch_size=4, drop_last=True)
    test_loader = DataLoader(tst, batch_size=4)

    model = ResNetDWT(Bottleneck, [1, 1, 1, 1], None, num_classes=7)
    opt = torch.optim.SGD(model.parameters(), lr=1e-3, momentum=0.9)
    args = _args(num_iters=2, check_acc_step=100)
    ckpt_path = str(tmp_path / "oh.pt")
    acc = train_infinite_collect_stats(
        args, model, torch.device("cpu"), src_loader, tgt_loader, opt, 0.1,
        test_loader, checkpoint_path=ckpt_path, checkpoint_every=1)
    assert 0.0 <= acc <= 100.0
    import os
    assert os.path.exists(ckpt_path)


def test_eval_pass_collect_stats_updates_buffers():
    torch.manual_seed(0)
    model = ResNetDWT(Bottleneck, [1, 1, 1, 1], None, num_classes=5)
    tst = SyntheticOfficeHome(4, num_classes=5, img_size=64, seed=3)
    loader = DataLoader(tst, batch_size=2)
    before = model.bnt1.wh.running_mean.clone()
    eval_pass_collect_stats(_args(), model, torch.device("cpu"), loader, passes=1)
    assert not torch.allclose(before, model.bnt1.wh.running_mean)


def test_entrypoint_digits_synthetic(capsys):
    from usps_mnist import main
    main(["--synthetic", "--synthetic_size", "64", "--epochs", "1",
          "--group_size", "4", "--num_workers", "0", "--test_batch_size", "16",
          "--log_interval", "1"])
    out = capsys.readouterr().out
    assert "Train Epoch" in out and "Test set" in out


def test_entrypoint_digits_mec_synthetic(capsys):
    from usps_mnist import main
    main(["--synthetic", "--synthetic_size", "32", "--epochs", "1",
          "--group_size", "4", "--loss", "mec", "--num_workers", "0",
          "--test_batch_size", "16", "--log_interval", "1"])
    assert "Test set" in capsys.readouterr().out


def test_jsonl_logger(tmp_path):
    import json
    from dwt_amd.engine.meters import JsonlLogger, AverageMeter, ThroughputMeter
    path = str(tmp_path / "m.jsonl")
    lg = JsonlLogger(path, rank=0)
    lg.log(kind="train", step=1, loss=0.5)
    lg.log(kind="test", acc=42.0)
    lg.close()
    rows = [json.loads(l) for l in open(path)]
    assert rows[0]["kind"] == "train" and rows[1]["acc"] == 42.0
    # rank != 0 writes nothing
    lg2 = JsonlLogger(str(tmp_path / "n.jsonl"), rank=1)
    lg2.log(kind="x")
    lg2.close()
    import os
    assert not os.path.exists(str(tmp_path / "n.jsonl"))
    m = AverageMeter(); m.update(2.0, 3); m.update(5.0)
    assert abs(m.avg - 11.0 / 4) < 1e-9
    tp = ThroughputMeter(); assert tp.tick(10) is None; assert tp.tick(10) > 0


@pytest.mark.slow
def test_officehome_sweep_single_pair(tmp_path):
    import json
    import subprocess, sys, os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = str(tmp_path / "sweep.json")
    r = subprocess.run(
        [sys.executable, "benchmarks/officehome_sweep.py", "--synthetic",
         "--synthetic_size", "24", "--num_iters", "1", "--out", out,
         "--pairs", "Art:Clipart", "--source_batch_size", "4",
         "--test_batch_size", "8", "--num_workers", "0", "--check_acc_step",
         "100", "--log_interval", "1", "--stats_passes", "1"],
        cwd=repo, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-1500:]
    data = json.load(open(out))
    assert "Art->Clipart" in data and "average" in data


def test_officehome_resume_fast_forwards_scheduler():
    """start_iter fast-forwards the MultiStepLR past its milestone."""
    from dwt_amd.ops.optim import FusedSGD
    src = SyntheticOfficeHome(8, num_classes=5, img_size=64, seed=1)
    tgt = SyntheticOfficeHome(8, num_classes=5, img_size=64, transform_aug=True, seed=2)
    tst = SyntheticOfficeHome(4, num_classes=5, img_size=64, seed=2)
    src_loader = DataLoader(src, batch_size=4, drop_last=True)
    tgt_loader = DataLoader(tgt, batch_size=4, drop_last=True)
    test_loader = DataLoader(tst, batch_size=4)
    model = ResNetDWT(Bottleneck, [1, 1, 1, 1], None, num_classes=5)
    opt = FusedSGD(model.parameters(), lr=1.0, momentum=0.0)
    args = _args(num_iters=6002, check_acc_step=10**9, stats_passes=0)
    # start past the 6000-iter milestone: lr must arrive decayed by 0.1
    train_infinite_collect_stats(args, model, torch.device("cpu"), src_loader,
                                 tgt_loader, opt, 0.1, test_loader,
                                 start_iter=6001, stats_passes=0)
    assert abs(opt.param_groups[0]["lr"] - 0.1) < 1e-9


def test_emergency_checkpoint_on_failure(tmp_path):
    from dwt_amd.models import checkpoint as ckptmod
    src = SyntheticOfficeHome(8, num_classes=5, img_size=64, seed=1)
    tgt = SyntheticOfficeHome(8, num_classes=5, img_size=64, transform_aug=True, seed=2)
    tst = SyntheticOfficeHome(4, num_classes=5, img_size=64, seed=2)
    src_loader = DataLoader(src, batch_size=4, drop_last=True)
    tgt_loader = DataLoader(tgt, batch_size=4, drop_last=True)
    test_loader = DataLoader(tst, batch_size=4)
    model = ResNetDWT(Bottleneck, [1, 1, 1, 1], None, num_classes=5)
    opt = torch.optim.SGD(model.parameters(), lr=1e-3)
    path = str(tmp_path / "oh.pt")

    class Boom(RuntimeError):
        pass

    calls = {"n": 0}
    orig_forward = model.forward

    def exploding(x):
        calls["n"] += 1
        if calls["n"] >= 3:
            raise Boom("injected fault")
        return orig_forward(x)

    model.forward = exploding
    args = _args(num_iters=10, check_acc_step=10**9)
    with pytest.raises(Boom):
        train_infinite_collect_stats(args, model, torch.device("cpu"),
                                     src_loader, tgt_loader, opt, 0.1,
                                     test_loader, checkpoint_path=path)
    import os
    assert os.path.exists(path + ".emergency")
    model.forward = orig_forward
    model2 = ResNetDWT(Bottleneck, [1, 1, 1, 1], None, num_classes=5)
    it = ckptmod.load_training_state(path + ".emergency", model2)
    assert it == 2  # failed during iteration 2's forward


def test_officehome_entrypoint_gpu_augment_cpu():
    """--gpu_augment: the duplicate MEC view is built on-device by the
    engine (batched flip+affine) instead of in the loader workers."""
    import subprocess, sys, os as _os
    repo = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "resnet50_dwt_mec_officehome.py", "--synthetic",
         "--synthetic_size", "64", "--num_iters", "3", "--check_acc_step",
         "100", "--source_batch_size", "4", "--test_batch_size", "8",
         "--num_workers", "0", "--log_interval", "1", "--img_crop_size",
         "64", "--gpu_augment", "--stats_passes", "1"],
        cwd=repo, capture_output=True, text=True, timeout=900)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "Train Iter" in r.stdout
    assert "Test set" in r.stdout
