import json

import pytest

from distributed_reinforcement_learning_amd.config import (
    Config, check_properties, load_config,
)


def _block(**over):
    d = {
        "server_ip": "127.0.0.1", "server_port": 8000, "num_actors": 2,
        "env": ["E", "E"], "available_action": [4, 4],
        "batch_size": 8, "queue_size": 16, "discount_factor": 0.99,
        "model_input": [84, 84, 4], "model_output": 4,
        "reward_clipping": "abs_one", "trajectory": 20, "lstm_size": 256,
        "start_learning_rate": 6e-4, "end_learning_rate": 0.0,
        "learning_frame": 10 ** 9, "entropy_coef": 0.05,
        "baseline_loss_coef": 1.0, "gradient_clip_norm": 40.0,
    }
    d.update(over)
    return d


def test_load_config(tmp_path):
    p = tmp_path / "config.json"
    p.write_text(json.dumps({"impala": _block()}))
    cfg = load_config(str(p), "impala")
    assert cfg.batch_size == 8
    assert cfg.trajectory == 20
    assert cfg.model_input == [84, 84, 4]
    assert cfg["num_actors"] == 2  # raw access


def test_check_properties_rejects_bad_action_count():
    with pytest.raises(ValueError):
        check_properties(_block(available_action=[4, 99]))


def test_check_properties_rejects_length_mismatch():
    with pytest.raises(ValueError):
        check_properties(_block(env=["E"]))


def test_check_properties_rejects_unknown_clipping():
    with pytest.raises(ValueError):
        check_properties(_block(reward_clipping="bogus"))


def test_repo_config_blocks_load():
    from distributed_reinforcement_learning_amd.config import default_config_path
    for algo in ("a3c", "impala", "apex", "r2d2"):
        cfg = load_config(default_config_path(), algo)
        assert cfg.num_actors >= 1
