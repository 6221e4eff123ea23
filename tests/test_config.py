"""Config plumbing: YAML presets, CLI precedence, labels."""
import argparse

import pytest

from tensorflow_web_deploy_amd.conf.config import (Config, add_config_args,
                                                   config_from_args,
                                                   load_config)


def test_cli_overrides_yaml(tmp_path):
    y = tmp_path / "c.yaml"
    y.write_text("model_name: resnet50\nmax_batch: 32\nport: 6000\n")
    p = argparse.ArgumentParser()
    add_config_args(p)
    args = p.parse_args(["--config", str(y), "--max-batch", "64"])
    cfg = config_from_args(args)
    assert cfg.model_name == "resnet50"
    assert cfg.max_batch == 64      # CLI wins
    assert cfg.port == 6000         # YAML survives


def test_unknown_yaml_key_rejected(tmp_path):
    y = tmp_path / "c.yaml"
    y.write_text("model_name: lenet\nnot_a_field: 1\n")
    with pytest.raises(Exception):
        load_config(str(y))


def test_labels_default_and_file(tmp_path):
    from tensorflow_web_deploy_amd.server.labels import load_labels
    labs = load_labels(None, 7)
    assert len(labs) == 7
    f = tmp_path / "l.txt"
    f.write_text("cat\ndog\nfish\n")
    labs = load_labels(str(f), 5)
    assert labs[0] == "cat" and labs[2] == "fish"
    assert len(labs) == 5  # padded to num_classes
