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


def test_reference_key_aliases(tmp_path):
    import os
    import yaml
    from neuronx_distributed_training_amd.utils.config import load_config

    p = os.path.join(str(tmp_path), "ref.yaml")
    yaml.safe_dump(
        {
            "trainer": {"max_steps": 5, "gradient_clip_val": 1.0},
            "data": {"global_batch_size": 8, "micro_batch_size": 2},
            "model": {"encoder_seq_length": 64,
                      "activations_checkpoint_granularity": "selective"},
        },
        open(p, "w"),
    )
    cfg = load_config(p)
    assert cfg["model"]["grad_clip"] == 1.0
    assert cfg["model"]["activation_checkpoint"] == "selective"
    assert cfg["data"]["seq_length"] == 64
