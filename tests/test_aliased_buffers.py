"""Bound the one intentional semantic deviation from the reference.

The reference builds each norm site's three branches (bns*/bnt*/bnt*_aug)
around the SAME running-stat tensor objects and updates them in place
(resnet50_dwt_mec_officehome.py:74-90 + utils/whitening.py:58-59), so the
branches share one set of running statistics.  This framework keeps separate
per-branch buffers (clean semantics) and relies on the same
`eval_pass_collect_stats` re-estimation the reference runs before its final
test (resnet50_dwt_mec_officehome.py:380-389,443).

These tests reproduce the reference's aliased-buffer dynamics on an
oracle-level copy of the model and assert:

1. TRAINING dynamics are identical (training mode uses batch statistics, so
   weights/losses cannot depend on the aliasing) — the deviation is confined
   to the running buffers.
2. MID-training eval outputs DO differ (the deviation is real and measurable
   — this guards the test itself against vacuity).
3. After the re-estimation pass, final eval outputs of the two variants
   converge (the deviation is benign for the metric the reference reports).
"""
import copy

import pytest
import torch

from dwt_amd.models import Bottleneck, ResNetDWT


BRANCH_SITES = [
    ("bns1", "bnt1", "bnt1_aug"),
    ("bns2", "bnt2", "bnt2_aug"),
    ("bns3", "bnt3", "bnt3_aug"),
    ("downsample_bns", "downsample_bnt", "downsample_bnt_aug"),
]


def alias_running_buffers(model):
    """Make every norm site's three branches share ONE set of running-stat
    tensors, exactly as the reference constructs them."""
    n_sites = 0
    for m in model.modules():
        for src_name, *tgt_names in BRANCH_SITES:
            if not hasattr(m, src_name):
                continue
            src = getattr(m, src_name)
            for tname in tgt_names:
                tgt = getattr(m, tname)
                if hasattr(src, "wh"):  # whitening site
                    tgt.wh.running_mean = src.wh.running_mean
                    tgt.wh.running_variance = src.wh.running_variance
                else:  # domain BN site
                    tgt.running_mean = src.running_mean
                    tgt.running_var = src.running_var
            n_sites += 1
    assert n_sites > 0
    return model


def _make_model(seed=0):
    torch.manual_seed(seed)
    return ResNetDWT(Bottleneck, [1, 1, 1, 1], None, num_classes=5,
                     group_size=4).train()


def _train_steps(model, n_steps=4, b=2, lr=0.05):
    torch.manual_seed(42)
    opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=0.9)
    losses = []
    for _ in range(n_steps):
        data = torch.randn(3 * b, 3, 32, 32)
        y = torch.randint(0, 5, (b,))
        opt.zero_grad()
        out = model(data)
        s, t, td = torch.split(out, b, dim=0)
        loss = torch.nn.functional.cross_entropy(s, y) + 0.1 * (t - td).pow(2).mean()
        loss.backward()
        opt.step()
        losses.append(loss.item())
    return losses


def _reestimate(model, batches):
    """The reference's eval_pass_collect_stats: train-mode no-grad passes
    with the target batch tripled into all three streams."""
    model.train()
    with torch.no_grad():
        for _ in range(10):
            for data in batches:
                model(torch.cat((data, data, data), dim=0))


def test_training_dynamics_independent_of_aliasing():
    """Training-mode forward uses batch stats, so aliasing the running
    buffers must not change losses or learned weights."""
    m_sep = _make_model()
    m_ali = alias_running_buffers(_make_model())
    l_sep = _train_steps(m_sep)
    l_ali = _train_steps(m_ali)
    assert l_sep == pytest.approx(l_ali, rel=1e-6)
    for (n, p), (_, q) in zip(m_sep.named_parameters(), m_ali.named_parameters()):
        assert torch.allclose(p, q, atol=1e-7), n


def test_aliasing_deviation_real_then_benign_after_reestimation():
    m_sep = _make_model()
    m_ali = alias_running_buffers(_make_model())
    _train_steps(m_sep)
    _train_steps(m_ali)

    torch.manual_seed(7)
    eval_x = torch.randn(4, 3, 32, 32)

    # (2) mid-training eval: deviation must be measurable (eval uses the
    # running buffers, which the two variants accumulated differently)
    m_sep.eval(); m_ali.eval()
    with torch.no_grad():
        out_sep_mid = m_sep(eval_x)
        out_ali_mid = m_ali(eval_x)
    mid_gap = (out_sep_mid - out_ali_mid).abs().max().item()
    assert mid_gap > 1e-5, "aliased and separate buffers agreed mid-training;" \
                           " the deviation this test bounds does not exist"

    # (3) after the reference's target-stats re-estimation, both variants'
    # buffers converge to target statistics and final eval outputs agree
    # Enough batches that the EMA's memory of the pre-re-estimation buffers
    # vanishes (0.9^(10*16) ~ 5e-8; the aliased variant's effective per-batch
    # momentum is 0.271 because the shared buffer absorbs all three branch
    # updates, so its memory vanishes even faster).  What CANNOT vanish is
    # the difference between the two variants' EMA mixture weights over the
    # same noisy per-batch statistics — that is a stats-estimation-noise
    # effect, not a semantic one, so the bound below compares the
    # aliased-vs-separate gap against the noise floor measured by
    # re-estimating the SAME (separate-buffer) model on the same batches in
    # a different order.
    torch.manual_seed(11)
    target_batches = [torch.randn(4, 3, 32, 32) for _ in range(16)]
    m_sep2 = copy.deepcopy(m_sep)
    _reestimate(m_sep, target_batches)
    _reestimate(m_sep2, list(reversed(target_batches)))
    _reestimate(m_ali, target_batches)
    m_sep.eval(); m_sep2.eval(); m_ali.eval()
    with torch.no_grad():
        out_sep = m_sep(eval_x)
        out_sep2 = m_sep2(eval_x)
        out_ali = m_ali(eval_x)
    noise_floor = (out_sep - out_sep2).abs().max().item()
    final_gap = (out_sep - out_ali).abs().max().item()
    assert final_gap < max(5.0 * noise_floor, 1e-4), (final_gap, noise_floor)
    # and the re-estimation really closed the gap, not just shrank it a bit
    assert final_gap < mid_gap * 1e-3, (final_gap, mid_gap)
    # prediction-level agreement (the metric the reference reports is top-1)
    assert (out_sep.argmax(1) == out_ali.argmax(1)).all()
