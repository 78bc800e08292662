"""Schedulers, loggers, timings, checkpoints, config CLI."""

import json
import os

import pytest
import torch

from scalerl_amd.config import DQNArguments, ImpalaArguments, parse_cli
from scalerl_amd.utils import (LinearDecayScheduler, MultiStepScheduler,
                               PiecewiseScheduler, Timer, Timings)
from scalerl_amd.utils.checkpoint import (load_agent_checkpoint,
                                          load_checkpoint,
                                          save_agent_checkpoint,
                                          save_checkpoint)
from scalerl_amd.utils.loggers import JsonlLogger, make_logger


def test_linear_decay():
    s = LinearDecayScheduler(1.0, 0.1, 100)
    assert s.value(0) == 1.0
    assert s.value(50) == pytest.approx(0.55)
    assert s.value(1000) == pytest.approx(0.1)


def test_piecewise():
    s = PiecewiseScheduler([(0, 0.0), (10, 1.0), (20, 0.0)])
    assert s.value(5) == pytest.approx(0.5)
    assert s.value(15) == pytest.approx(0.5)
    assert s.value(99) == 0.0


def test_multistep():
    s = MultiStepScheduler(1.0, [10, 20], gamma=0.1)
    assert s.value(5) == 1.0
    assert s.value(15) == pytest.approx(0.1)
    assert s.value(25) == pytest.approx(0.01)


def test_timings_sections():
    t = Timings()
    t.reset()
    t.time("a")
    t.time("b")
    m = t.means()
    assert set(m) == {"a", "b"}
    assert "total" in t.summary()


def test_timer_context():
    with Timer() as t:
        sum(range(1000))
    assert t.elapsed >= 0


def test_jsonl_logger_roundtrip(tmp_path):
    lg = JsonlLogger(str(tmp_path))
    lg.log_train_data({"reward": 1.5}, step=10)
    lg.save_data(epoch=2, env_step=100, gradient_step=7)
    lg.close()
    lg2 = JsonlLogger(str(tmp_path))
    meta = lg2.restore_data()
    assert meta == {"epoch": 2, "env_step": 100, "gradient_step": 7}
    with open(lg2.path) as fh:
        recs = [json.loads(l) for l in fh]
    assert any(r["ns"] == "train" and r["reward"] == 1.5 for r in recs)


def test_make_logger_falls_back_to_jsonl(tmp_path):
    lg = make_logger("tensorboard", str(tmp_path))  # tb not installed here
    assert isinstance(lg, JsonlLogger)


def test_logger_interval_gating(tmp_path):
    lg = JsonlLogger(str(tmp_path), train_interval=10)
    lg.log_train_data({"x": 1}, step=0)
    lg.log_train_data({"x": 2}, step=5)   # gated out
    lg.log_train_data({"x": 3}, step=12)
    lg.close()
    with open(lg.path) as fh:
        recs = [json.loads(l) for l in fh]
    assert [r["x"] for r in recs] == [1.0, 3.0]


def test_checkpoint_formats(tmp_path):
    lin = torch.nn.Linear(3, 3)
    opt = torch.optim.Adam(lin.parameters())
    p = os.path.join(str(tmp_path), "model.tar")
    save_checkpoint(p, model=lin, optimizer=opt, hparam={"lr": 1e-3})
    ckpt = torch.load(p, map_location="cpu", weights_only=False)
    assert set(ckpt) >= {"model_state_dict", "optimizer_state_dict", "hparam"}
    lin2 = torch.nn.Linear(3, 3)
    load_checkpoint(p, model=lin2)
    torch.testing.assert_close(lin2.weight, lin.weight)

    p2 = os.path.join(str(tmp_path), "agent.pth")
    save_agent_checkpoint(p2, actor=lin, actor_target=lin2, optimizer=opt)
    ck2 = torch.load(p2, map_location="cpu", weights_only=False)
    assert set(ck2) >= {"actor_state_dict", "actor_target_state_dict",
                        "optimizer_state_dict"}


def test_parse_cli_types_and_defaults():
    a = parse_cli(DQNArguments, ["--learning-rate", "0.003",
                                 "--double-dqn", "false",
                                 "--batch-size", "128"])
    assert a.learning_rate == pytest.approx(0.003)
    assert a.double_dqn is False
    assert a.batch_size == 128
    b = parse_cli(ImpalaArguments, [])
    assert b.rollout_length == 80 and b.algo_name == "impala"
