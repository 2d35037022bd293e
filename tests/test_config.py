"""CLI surface parity: every reference flag exists with the reference's
default (distributed.py:25-102)."""

import pytest

from amdtrain.config import base_parser


def _defaults():
    p = base_parser("t")
    return vars(p.parse_args([]))


def test_reference_defaults():
    d = _defaults()
    # reference defaults (distributed.py:25-102)
    assert d["arch"] == "resnet18"
    assert d["workers"] == 4
    assert d["epochs"] == 90
    assert d["start_epoch"] == 0
    assert d["batch_size"] == 3200
    assert d["lr"] == 0.1
    assert d["momentum"] == 0.9
    assert d["weight_decay"] == 1e-4
    assert d["print_freq"] == 10
    assert d["evaluate"] is False
    assert d["pretrained"] is False
    assert d["seed"] is None


def test_short_flags():
    p = base_parser("t")
    d = vars(p.parse_args(["-a", "resnet50", "-b", "256", "-j", "8",
                           "-p", "5", "-e"]))
    assert d["arch"] == "resnet50"
    assert d["batch_size"] == 256
    assert d["workers"] == 8
    assert d["print_freq"] == 5
    assert d["evaluate"] is True


def test_arch_choices_validated():
    p = base_parser("t")
    with pytest.raises(SystemExit):
        p.parse_args(["-a", "vgg16"])


def test_wd_alias():
    p = base_parser("t")
    d = vars(p.parse_args(["--wd", "0.01"]))
    assert d["weight_decay"] == 0.01
    d = vars(p.parse_args(["--weight-decay", "0.02"]))
    assert d["weight_decay"] == 0.02


def test_local_rank_flag_on_ddp_entrypoints():
    from amdtrain.cli.distributed import parse_args
    a = parse_args(["--local_rank", "3"])
    assert a.local_rank == 3
    a = parse_args(["--local-rank", "2"])  # new-style spelling
    assert a.local_rank == 2


def test_start_sh_modules_exist():
    """Every `-m amdtrain.cli.X` in scripts/start.sh must be a real,
    importable entrypoint (guards doc rot in the canonical launch lines,
    reference start.sh:1-5)."""
    import importlib
    import os
    import re
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    text = open(os.path.join(root, "scripts", "start.sh")).read()
    mods = set(re.findall(r"-m (amdtrain\.cli\.\w+)", text))
    assert len(mods) == 6, mods
    for m in sorted(mods):
        mod = importlib.import_module(m)
        assert hasattr(mod, "main"), m
