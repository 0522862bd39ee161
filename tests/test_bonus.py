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
