"""CLI-surface parity with the reference entrypoints (SURVEY §5 config row:
'same CLI surface on the two entrypoints (contract!)')."""
import re

# flag lists transcribed from the reference argparse blocks
# (usps_mnist.py:332-348, resnet50_dwt_mec_officehome.py:499-518)
REF_DIGITS_FLAGS = [
    "--num_workers", "--source_batch_size", "--target_batch_size",
    "--test_batch_size", "--source", "--target", "--epochs", "--lr",
    "--sgd_momentum", "--running_momentum", "--lambda_entropy_loss",
    "--log_interval", "--seed", "--from_script", "--run", "--method",
    "--group_size",
]
REF_OFFICEHOME_FLAGS = [
    "--num_workers", "--source_batch_size", "--target_batch_size",
    "--test_batch_size", "--s_dset_path", "--t_dset_path", "--resnet_path",
    "--img_resize", "--img_crop_size", "--num_iters", "--check_acc_step",
    "--lr_change_step", "--lr", "--num_classes", "--sgd_momentum",
    "--running_momentum", "--lambda_mec_loss", "--log_interval", "--seed",
]


def _flags_of(parser):
    out = set()
    for a in parser._actions:
        out.update(a.option_strings)
    return out


def test_digits_cli_superset_of_reference():
    from usps_mnist import build_args
    import argparse
    # build the parser without parsing by reaching into build_args via a probe
    try:
        build_args(["--help"])
    except SystemExit:
        pass
    # direct approach: parse defaults and check attribute presence
    args = build_args([])
    for flag in REF_DIGITS_FLAGS:
        attr = flag.lstrip("-")
        assert hasattr(args, attr), f"missing reference flag {flag}"
    # reference defaults preserved
    assert args.source_batch_size == 32 and args.test_batch_size == 100
    assert args.epochs == 120 and args.lr == 0.001
    assert args.lambda_entropy_loss == 0.1 and args.group_size == 32
    assert args.source == "usps" and args.target == "mnist"


def test_officehome_cli_superset_of_reference():
    from resnet50_dwt_mec_officehome import build_args
    args = build_args([])
    for flag in REF_OFFICEHOME_FLAGS:
        attr = flag.lstrip("-")
        assert hasattr(args, attr), f"missing reference flag {flag}"
    assert args.source_batch_size == 18 and args.test_batch_size == 10
    assert args.num_iters == 10000 and args.check_acc_step == 100
    assert args.lr == 1e-2 and args.num_classes == 65
    assert args.lambda_mec_loss == 0.1 and args.img_crop_size == 224


def test_console_strings_match_reference_format(capsys):
    """The training/eval print lines keep the reference's exact formats
    (usps_mnist.py:306-308,323-325; resnet50...py:433-436,460-462)."""
    from usps_mnist import main
    main(["--synthetic", "--synthetic_size", "32", "--epochs", "1",
          "--group_size", "4", "--num_workers", "0", "--test_batch_size", "16",
          "--log_interval", "1"])
    out = capsys.readouterr().out
    assert re.search(
        r"Train Epoch: 0 \[\d+/\d+ \(\d+%\)\]\tClassification Loss: \d+\.\d{6} "
        r"\tEntropy Loss: \d+\.\d{6}", out), out[:400]
    assert re.search(
        r"Test set: Classification loss: \d+\.\d{4}, Accuracy: \d+/\d+ "
        r"\(\d+\.\d{2}%\)", out)
