"""BitFlip DQN + HER config (see ding/example/dqn_her.py for the pipeline)."""
from ding.utils import EasyDict

n_bits = 15

main_config = EasyDict(dict(
    exp_name='bitflip_her_dqn',
    env=dict(collector_env_num=4, evaluator_env_num=4, n_evaluator_episode=8, stop_value=0.9,
             n_bits=n_bits),
    policy=dict(
        cuda=True, nstep=1, discount_factor=0.98,
        model=dict(obs_shape=2 * n_bits, action_shape=n_bits, encoder_hidden_size_list=[256, 256]),
        learn=dict(update_per_collect=10, batch_size=128, learning_rate=5e-4, target_update_freq=500),
        collect=dict(n_episode=8, unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=500)),
        other=dict(eps=dict(type='exp', start=0.95, end=0.1, decay=100000),
                   replay_buffer=dict(replay_buffer_size=100000)),
    ),
    her=dict(her_strategy='future', her_replay_k=4),
))

create_config = EasyDict(dict(
    env=dict(type='bitflip', import_names=['dizoo.bitflip.envs.bitflip_env']),
    env_manager=dict(type='base'),
    policy=dict(type='dqn'),
))
