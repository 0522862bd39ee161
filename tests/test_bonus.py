"""Bonus Agent API: train/deploy/collect_data/batch_evaluate round-trip."""
import pytest

from ding.bonus import DQNAgent, PPOF, SACAgent


def test_dqn_agent(tmp_path):
    agent = DQNAgent(env_id='CartPole-v0', exp_name=str(tmp_path / 'agent'), seed=0)
    agent.train(step=200, max_train_iter=3)
    ret = agent.deploy(max_episode_steps=50)
    assert isinstance(ret.eval_value, float)
    agent.collect_data(n_sample=16, save_data_path=str(tmp_path / 'demo.pkl'))
    ev = agent.batch_evaluate()
    assert isinstance(ev.eval_value, float)


def test_ppof_agent(tmp_path):
    agent = PPOF(env_id='CartPole-v0', exp_name=str(tmp_path / 'ppof'), seed=0)
    agent.train(step=300, max_train_iter=2)
    ret = agent.deploy(max_episode_steps=50)
    assert isinstance(ret.eval_value, float)


def test_sac_agent(tmp_path):
    agent = SACAgent(env_id='Pendulum-v1', exp_name=str(tmp_path / 'sac'), seed=0)
    agent.train(step=200, max_train_iter=2)
    ret = agent.deploy(max_episode_steps=50)
    assert isinstance(ret.eval_value, float)


def test_bonus_example_presets():
    """Agents pick up tuned presets from ding/config/example/<ALGO>/."""
    from ding.config.example import get_example_config, list_example_configs
    all_presets = list_example_configs()
    assert len(all_presets) >= 25, all_presets
    cfg = get_example_config('DQN', 'LunarLander-v2')
    assert cfg.env.stop_value == 200 and cfg.policy.model.obs_shape == 8
    assert get_example_config('DQN', 'NoSuchEnv-v0') is None

    from ding.bonus import DQNAgent
    agent = DQNAgent(env_id='LunarLander-v2', exp_name='exp/test_bonus_ll_dqn')
    assert agent.main_config.policy.model.action_shape == 4
    assert agent.create_config.env.type == 'lunarlander'
    # 1-iter training still works through the preset
    agent.train(max_train_iter=1, collector_env_num=2, evaluator_env_num=1)


def test_bonus_sac_preset_smoke():
    from ding.bonus import SACAgent
    agent = SACAgent(env_id='LunarLanderContinuous-v2', exp_name='exp/test_bonus_llc_sac',
                     cfg=dict(policy=dict(cuda=False, random_collect_size=16,
                                          learn=dict(batch_size=8),
                                          other=dict(replay_buffer=dict(replay_buffer_size=1000)))))
    agent.train(max_train_iter=1, collector_env_num=2, evaluator_env_num=1)
