"""Shipped example configs must stay runnable end-to-end via the CLI
(scaled down by CLI overrides so CPU finishes in seconds)."""
import json

import pytest

from gymfx_amd.main import main as cli_main


@pytest.mark.parametrize("cfg_file,overrides", [
    ("examples/config/buy_hold.json", []),
    ("examples/config/random_driver.json", []),
    ("examples/config/train_ppo_mlp.json",
     ["--n_envs", "16", "--synthetic_rows", "2000", "--train_updates", "2",
      "--rollout_steps", "16", "--minibatches", "2", "--ppo_epochs", "1",
      "--hidden_size", "16"]),
    ("examples/config/train_ppo_lstm.json",
     ["--n_envs", "16", "--synthetic_rows", "2000", "--train_updates", "2",
      "--rollout_steps", "16", "--minibatches", "2", "--ppo_epochs", "1",
      "--hidden_size", "16", "--bptt_len", "4"]),
])
def test_example_config_runs(tmp_path, cfg_file, overrides):
    res = tmp_path / "r.json"
    cli_main(["--load_config", cfg_file, "--quiet_mode", "true",
              "--results_file", str(res), "--save_config", "",
              "--device", "cpu",
              "--checkpoint_file", str(tmp_path / "c.pt"),
              *overrides])
    out = json.loads(res.read_text())
    assert out, cfg_file
