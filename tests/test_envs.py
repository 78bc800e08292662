"""Env layer: wrappers, vectorization, async workers, dict protocol."""

import numpy as np
import pytest
import torch

from scalerl_amd.envs import (CartPoleEnv, SyntheticAtariEnv, TorchEnvWrapper,
                              make_env, make_vect_envs)
from scalerl_amd.envs.async_vec_env import AsyncVectorEnv
from scalerl_amd.envs.atari_wrappers import (ClipRewardEnv, FrameStack,
                                             MaxAndSkipEnv, WarpFrame,
                                             wrap_deepmind)
from scalerl_amd.envs.synthetic import SyntheticAtariVecEnv


def test_cartpole_api():
    env = CartPoleEnv(seed=0)
    obs, info = env.reset(seed=0)
    assert obs.shape == (4,)
    for _ in range(10):
        obs, r, term, trunc, _ = env.step(env.action_space.sample())
        assert r == 1.0
        if term or trunc:
            break


def test_cartpole_deterministic_with_seed():
    a = CartPoleEnv()
    b = CartPoleEnv()
    oa, _ = a.reset(seed=7)
    ob, _ = b.reset(seed=7)
    np.testing.assert_array_equal(oa, ob)
    for _ in range(5):
        oa, *_ = a.step(1)
        ob, *_ = b.step(1)
        np.testing.assert_array_equal(oa, ob)


def test_synthetic_vec_matches_scalar_reward_rule():
    v = SyntheticAtariVecEnv(4, seed=0)
    obs = v.reset()
    assert obs.shape == (4, 4, 84, 84) and obs.dtype == np.uint8
    correct = v._state % v.num_actions
    _, r, d = v.step(correct)
    assert (r == 1.0).all()
    _, r2, _ = v.step((correct + 1) % v.num_actions)  # states moved; mixed
    assert r2.shape == (4,)


def test_wrappers_compose():
    env = wrap_deepmind(SyntheticAtariEnv(seed=0), episode_life=False,
                        noop_max=3, skip=2, warp=False, frame_stack=2,
                        clip_rewards=True)
    obs, _ = env.reset(seed=0)
    assert obs.shape[0] == 2  # frame stack over the raw (4,84,84) obs
    _, r, *_ = env.step(0)
    assert r in (-1.0, 0.0, 1.0)


def test_warp_frame_shapes():
    class Rgb210(SyntheticAtariEnv):
        def __init__(self):
            super().__init__(frame_shape=(210, 160, 3), seed=0)
    env = WarpFrame(Rgb210())
    obs, _ = env.reset()
    assert obs.shape == (84, 84) and obs.dtype == np.uint8


def test_sync_vec_env_autoreset():
    v = make_vect_envs("CartPole-v1", 3, seed=0)
    obs = v.reset(seed=0)
    assert obs.shape == (3, 4)
    for _ in range(600):  # long enough to hit terminations → autoreset
        obs, r, d = v.step(np.ones(3, dtype=np.int64))
    assert obs.shape == (3, 4)
    v.close()


def test_async_vec_env_roundtrip():
    v = AsyncVectorEnv([lambda: CartPoleEnv(seed=i) for i in range(3)])
    try:
        obs = v.reset(seed=0)
        assert obs.shape == (3, 4)
        obs, r, d = v.step([0, 1, 0])
        assert r.shape == (3,) and d.shape == (3,)
    finally:
        v.close()


def test_async_vec_env_worker_crash_surfaces():
    class Boom(CartPoleEnv):
        def step(self, action):
            raise ValueError("boom")
    v = AsyncVectorEnv([lambda: Boom()])
    try:
        v.reset(seed=0)
        with pytest.raises((RuntimeError, TimeoutError)):
            v.step([0])
            v.step_wait()
    finally:
        v.close(terminate=True)


def test_torch_env_wrapper_dict_protocol():
    w = TorchEnvWrapper(make_env("CartPole-v1", seed=0))
    out = w.initial(seed=0)
    assert set(out) == {"obs", "reward", "done", "episode_return",
                        "episode_step", "last_action"}
    assert out["obs"].shape == (1, 1, 4)
    assert bool(out["done"])
    out = w.step(torch.tensor(1))
    assert out["episode_step"].item() == 1
    assert out["reward"].item() == 1.0


def test_record_episode_statistics_contract():
    """gymnasium RecordEpisodeStatistics dict contract on terminal steps
    (reference make_gym_env parity, gym_env.py:6-33)."""
    from scalerl_amd.envs.registry import make_gym_env
    env = make_gym_env("CartPole-v1", seed=0)
    obs, _ = env.reset(seed=0)
    done_info = None
    for _ in range(501):
        obs, r, term, trunc, info = env.step(env.action_space.sample())
        if term or trunc:
            done_info = info
            break
    assert done_info is not None and "episode" in done_info
    ep = done_info["episode"]
    assert ep["l"] >= 1 and ep["r"] == ep["l"]  # CartPole reward = 1/step
    env.close()


def test_record_video_writes_gif(tmp_path):
    """RecordVideo captures the triggered episode's frames to a GIF."""
    from scalerl_amd.envs.registry import make_gym_env
    env = make_gym_env("synthetic-atari", seed=0, capture_video=True,
                       save_video_dir=str(tmp_path), save_video_name="t")
    env.reset(seed=0)  # episode 0 triggers (0^3 == 0)
    for _ in range(40):
        obs, r, term, trunc, info = env.step(env.action_space.sample())
        if term or trunc:
            break
    env.close()
    import glob
    files = glob.glob(str(tmp_path / "t" / "*.gif"))
    assert files, "no GIF written"
    from PIL import Image
    im = Image.open(files[0])
    assert im.n_frames >= 2


def _have_ale() -> bool:
    try:
        import gymnasium  # noqa: F401
        import ale_py  # noqa: F401
        return True
    except ImportError:
        return False


@pytest.mark.skipif(not _have_ale(), reason="gymnasium/ALE not installed "
                    "in this image (no network); runs where they exist")
def test_real_ale_deepmind_stack_smoke():
    """Compose the DeepMind wrapper stack over a real ALE env through
    GymAdapter (reference wrap_deepmind, atari_wrapper.py:277-311)."""
    from scalerl_amd.envs.registry import make_env
    env = make_env("ALE/Pong-v5", seed=0, deepmind_wrap=True)
    obs, _ = env.reset(seed=0)
    assert obs.shape == (4, 84, 84) and obs.dtype.name == "uint8"
    for _ in range(20):
        obs, r, term, trunc, info = env.step(env.action_space.sample())
        assert obs.shape == (4, 84, 84)
        if term or trunc:
            break
    env.close()
