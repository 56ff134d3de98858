"""Every shipped recipe YAML must load and validate."""

import glob
import os

import pytest

from neuronx_distributed_training_amd.utils.config import load_config

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CONFIGS = sorted(glob.glob(os.path.join(REPO, "examples", "conf", "*.yaml")))


@pytest.mark.parametrize("path", CONFIGS, ids=[os.path.basename(p) for p in CONFIGS])
def test_config_loads(path):
    cfg = load_config(path)
    assert "model" in cfg and "data" in cfg
    d = cfg["data"]
    assert int(d["global_batch_size"]) % int(d["micro_batch_size"]) == 0
