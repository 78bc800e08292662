"""Example entry points run end-to-end with tiny configs (subprocess —
exactly as a user would invoke them)."""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(script, args, timeout=420):
    r = subprocess.run([sys.executable, os.path.join(REPO, "examples", script)]
                       + args, capture_output=True, text=True,
                       timeout=timeout, cwd=REPO)
    assert r.returncode == 0, (script, r.stdout[-800:], r.stderr[-2000:])
    return r


def test_train_dqn_example(tmp_path):
    _run("train_dqn.py",
         ["--env-id", "CartPole-v1", "--num-envs", "2",
          "--max-train-steps", "400", "--warmup-learn-steps", "100",
          "--buffer-size", "1000", "--batch-size", "32",
          "--eval-episodes", "1", "--work-dir", str(tmp_path),
          "--save-model", "false", "--device", "cpu"])


def test_train_a3c_example(tmp_path):
    _run("train_a3c.py",
         ["--env-id", "CartPole-v1", "--num-workers", "2",
          "--max-train-steps", "600", "--rollout-steps", "16",
          "--eval-episodes", "1", "--work-dir", str(tmp_path),
          "--save-model", "false"])


def test_train_apex_example(tmp_path):
    _run("train_apex.py",
         ["--num-actors", "2", "--envs-per-actor", "4",
          "--buffer-size", "2048", "--batch-size", "32",
          "--warmup-learn-steps", "128", "--max-train-steps", "1500",
          "--learner-update-times", "1", "--device", "cpu",
          "--work-dir", str(tmp_path), "--save-model", "false"])


def test_train_ddppo_example(tmp_path):
    _run("train_ddppo.py",
         ["--rollout-length", "6", "--num-envs", "2",
          "--ppo-epochs", "1", "--num-minibatches", "1",
          "--max-train-steps", "12", "--device", "cpu",
          "--work-dir", str(tmp_path), "--save-model", "false"])


def test_train_impala_example(tmp_path):
    _run("train_impala.py",
         ["--rollout-length", "8", "--batch-size", "8",
          "--envs-per-actor", "4", "--num-actors", "2",
          "--total-steps", "128", "--device", "cpu", "--dtype", "fp32",
          "--output-dir", str(tmp_path),
          "--checkpoint-interval-s", "100000"])
