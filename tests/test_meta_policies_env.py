"""MetaExample packing, meta policies, run_meta_env, pose_env MAML loop."""

import numpy as np
import pytest
import torch

from tensor2robot_amd.data import example as example_mod
from tensor2robot_amd.meta_learning import meta_example
from tensor2robot_amd.meta_learning import meta_policies
from tensor2robot_amd.meta_learning import run_meta_env as run_meta_env_mod
from tensor2robot_amd.predictors import checkpoint_predictor
from tensor2robot_amd.research.pose_env import pose_env
from tensor2robot_amd.research.pose_env import pose_env_maml_models
from tensor2robot_amd.research.pose_env import pose_env_models


def test_make_meta_example_prefixes():
  cond = [{"state": [1.0, 2.0]}, {"state": [3.0, 4.0]}]
  inf = [{"state": [5.0, 6.0]}]
  meta = meta_example.make_meta_example(cond, inf)
  assert set(meta.keys()) == {"condition_ep0/state", "condition_ep1/state",
                              "inference_ep0/state"}
  # Round-trips through the native proto codec.
  data = example_mod.encode_example(meta)
  decoded = example_mod.decode_example(data)
  assert set(decoded.keys()) == set(meta.keys())


def test_make_meta_sequence_example():
  cond = [({"task": [1]}, {"obs": [[1.0], [2.0]]})]
  inf = [({"task": [2]}, {"obs": [[3.0]]})]
  context, lists = meta_example.make_meta_example(cond, inf)
  assert "condition_ep0/task" in context
  assert "inference_ep0/obs" in lists


def _maml_model_and_predictor():
  base = pose_env_models.PoseEnvRegressionModel()
  model = pose_env_maml_models.PoseEnvRegressionModelMAML(
      base_model=base, device_type="cpu", compute_dtype="float32",
      num_inner_loop_steps=1, inner_learning_rate=1e-3)
  predictor = checkpoint_predictor.CheckpointPredictor(t2r_model=model)
  predictor.init_randomly()
  return model, predictor


def test_maml_regression_policy_select_action():
  model, predictor = _maml_model_and_predictor()
  policy = meta_policies.MAMLRegressionPolicy(t2r_model=model,
                                              predictor=predictor)
  policy.reset_task()
  state = (np.random.RandomState(0).rand(64, 64, 3) * 255).astype(np.uint8)
  action = policy.SelectAction(state, None, 0)
  assert action.shape == (2,)
  # Adapt on a fake episode and act again -> still valid, changed output.
  episode = [[(state, np.array([0.1, -0.2], np.float32), 1.0, state,
               True, {})]]
  policy.adapt(episode)
  action2, debug = policy.sample_action(state)
  assert action2.shape == (2,)
  assert debug["is_demo"] is False


def test_run_meta_env_pose_toy():
  model, predictor = _maml_model_and_predictor()
  policy = meta_policies.MAMLRegressionPolicy(t2r_model=model,
                                              predictor=predictor)
  env = pose_env.PoseToyEnv(seed=3)
  step_means = run_meta_env_mod.run_meta_env(
      env, policy=policy, num_tasks=2, num_adaptations_per_task=2,
      num_episodes_per_adaptation=1, num_demos=0,
      max_episode_steps=3)
  assert len(step_means) == 2
  assert all(np.isfinite(m) for m in step_means)


def test_run_meta_env_with_demo_policy(tmp_path):
  model, predictor = _maml_model_and_predictor()
  policy = meta_policies.MAMLRegressionPolicy(t2r_model=model,
                                              predictor=predictor)
  env = pose_env.PoseToyEnv(seed=4)

  class DemoPolicy:
    def __init__(self, env):
      self._steps = 0

    def sample_action(self, obs, explore_prob):
      self._steps += 1
      if self._steps > 2:
        return None, {}
      return np.array([0.0, 0.0], np.float32), {}

  step_means = run_meta_env_mod.run_meta_env(
      env, policy=policy, demo_policy_cls=DemoPolicy,
      num_tasks=1, num_adaptations_per_task=1,
      num_episodes_per_adaptation=1, num_demos=1,
      root_dir=str(tmp_path), max_episode_steps=2,
      break_after_one_task=True)
  assert len(step_means) == 1


def test_scheduled_exploration_maml_policy():
  model, predictor = _maml_model_and_predictor()
  policy = meta_policies.ScheduledExplorationMAMLRegressionPolicy(
      t2r_model=model, predictor=predictor, initial_sigma=0.0,
      final_sigma=0.0, seed=0)
  policy.reset_task()
  state = np.zeros((64, 64, 3), np.uint8)
  action, _ = policy.sample_action(state)
  assert action.shape == (2,)


def test_gym_env_adapters_both_apis():
  """envs.GymEnvAdapter maps classic-gym and gymnasium step/reset
  signatures onto the run_env contract (reference run_env.py:50-74)."""
  from tensor2robot_amd import envs as envs_mod

  class OldGym:
    def reset(self):
      return [0.0]

    def step(self, action):
      return [1.0], 2.0, False, {"k": 1}

  class NewGym:
    def reset(self):
      return [0.0], {"info": True}

    def step(self, action):
      return [1.0], 2.0, False, True, {"k": 2}

  for env_cls, key in ((OldGym, 1), (NewGym, 2)):
    ad = envs_mod.GymEnvAdapter(env_cls())
    assert ad.reset() == [0.0]
    obs, r, done, dbg = ad.step(0)
    assert obs == [1.0] and r == 2.0 and dbg["k"] == key
    assert done == (env_cls is NewGym)  # truncated=True in NewGym

  limited = envs_mod.TimeLimitWrapper(
      envs_mod.GymEnvAdapter(OldGym()), max_episode_steps=3)
  limited.reset()
  dones = [limited.step(0)[2] for _ in range(3)]
  assert dones == [False, False, True]


def test_run_env_with_adapter(tmp_path):
  """The episode runner drives an adapted env end to end."""
  import numpy as np
  from tensor2robot_amd import envs as envs_mod
  from tensor2robot_amd.research.dql_grasping_lib import run_env as re_mod

  class TinyGym:
    def __init__(self):
      self.t = 0

    def reset(self):
      self.t = 0
      return np.zeros(2, np.float32)

    def step(self, action):
      self.t += 1
      return (np.full(2, self.t, np.float32), 1.0, self.t >= 4, {})

  class RandomPolicy:
    def reset(self):
      pass

    def sample_action(self, obs, explore_prob):
      return np.zeros(1, np.float32), {}

  env = envs_mod.TimeLimitWrapper(envs_mod.GymEnvAdapter(TinyGym()),
                                  max_episode_steps=10)
  rewards = re_mod.run_env(env, policy=RandomPolicy(), num_episodes=2,
                           root_dir=str(tmp_path))
  assert rewards == [4.0, 4.0]
