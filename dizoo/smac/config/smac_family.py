"""Factory for the SMAC per-map config families. The reference ships ~50
hand-written files (dizoo/smac/config/smac_<map>_<algo>_config.py); here one
factory builds the same pairs on the cooperative-matrix env with each map's
agent count / action arity (no StarCraft II binary in this image).
"""
from ding.utils import EasyDict

# map -> (agent_num, action_dim)  (n_actions = 6 + n_enemies in real SMAC)
MAPS = {
    '3m': (3, 9),
    '8m': (8, 14),
    '2s3z': (5, 11),
    '3s5z': (8, 14),
    '5m6m': (5, 12),
    'MMM': (10, 16),
    'MMM2': (10, 18),
    '2c64zg': (2, 70),
    '3s5zvs3s6z': (8, 15),
    '10m11m': (10, 17),
    '25m': (25, 31),
    '27m30m': (27, 36),
    'corridor': (6, 30),
    '6h8z': (6, 14),
}

_OBS = 32


def _env_block(map_name: str) -> dict:
    agent_num, action_dim = MAPS[map_name]
    return dict(
        map_name=map_name,
        agent_num=agent_num,
        action_dim=action_dim,
        obs_dim=_OBS,
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=0.999 * 25,
    )


def _q_family(agent_num: int, action_dim: int, policy_type: str) -> dict:
    pol = dict(
        cuda=True,
        model=dict(
            agent_num=agent_num,
            obs_shape=_OBS,
            global_obs_shape=agent_num * action_dim,
            action_shape=action_dim,
            hidden_size_list=[64, 64],
            mixer=True,
        ),
        learn=dict(
            update_per_collect=20,
            batch_size=32,
            learning_rate=5e-4,
            clip_value=100,
            target_update_theta=0.008,
            discount_factor=0.99,
            double_q=False,
        ),
        collect=dict(n_sample=32, unroll_len=10, env_num=8),
        eval=dict(env_num=8, evaluator=dict(eval_freq=100, )),
        other=dict(
            eps=dict(type='exp', start=1, end=0.05, decay=50000),
            replay_buffer=dict(replay_buffer_size=5000, ),
        ),
    )
    if policy_type == 'wqmix':
        pol['learn']['alpha'] = 0.5
    if policy_type == 'coma':
        pol['model'] = dict(
            agent_num=agent_num,
            obs_shape=dict(agent_state=_OBS, global_state=agent_num * action_dim),
            action_shape=action_dim,
            actor_hidden_size_list=[64, 64],
        )
        pol['learn'].pop('clip_value', None)
        pol['learn'].pop('double_q', None)
        pol['learn']['td_lambda'] = 0.8
        pol['learn']['policy_weight'] = 0.001
        pol['learn']['value_weight'] = 1.0
        pol['learn']['entropy_weight'] = 0.01
    if policy_type == 'madqn':
        pol['nstep'] = 3
        pol['model']['mixer'] = False
        pol['model']['hidden_size_list'] = [256, 256]
    return pol


def _mappo(agent_num: int, action_dim: int) -> dict:
    return dict(
        cuda=True,
        multi_agent=True,
        action_space='discrete',
        model=dict(
            action_space='discrete',
            agent_num=agent_num,
            agent_obs_shape=_OBS,
            global_obs_shape=agent_num * action_dim,
            action_shape=action_dim,
        ),
        learn=dict(
            epoch_per_collect=5,
            batch_size=320,
            learning_rate=5e-4,
            value_weight=0.5,
            entropy_weight=0.01,
            clip_ratio=0.2,
            adv_norm=False,
            value_norm=True,
        ),
        collect=dict(n_sample=3200, unroll_len=1, discount_factor=0.99, gae_lambda=0.95, env_num=8),
        eval=dict(env_num=8, evaluator=dict(eval_freq=100, )),
    )


def _masac(agent_num: int, action_dim: int) -> dict:
    return dict(
        cuda=True,
        multi_agent=True,
        random_collect_size=0,
        model=dict(
            type='discrete_maqac',
            import_names=['ding.model.template.maqac'],
            agent_obs_shape=_OBS,
            global_obs_shape=agent_num * action_dim,
            action_shape=action_dim,
            twin_critic=True,
            actor_head_hidden_size=256,
            critic_head_hidden_size=256,
        ),
        learn=dict(
            update_per_collect=20,
            batch_size=64,
            learning_rate_q=5e-4,
            learning_rate_policy=5e-4,
            learning_rate_alpha=5e-5,
            target_theta=0.005,
            discount_factor=0.99,
            auto_alpha=False,
            log_space=True,
        ),
        collect=dict(n_sample=1600, unroll_len=1, env_num=8),
        eval=dict(env_num=8, evaluator=dict(eval_freq=100, )),
        other=dict(
            eps=dict(type='linear', start=1, end=0.05, decay=100000),
            replay_buffer=dict(replay_buffer_size=100000, ),
        ),
    )


def build_smac_config(map_name: str, algo: str):
    agent_num, action_dim = MAPS[map_name]
    if algo in ('qmix', 'wqmix', 'qtran', 'collaq', 'vdn', 'coma', 'madqn'):
        policy_type = 'qmix' if algo == 'vdn' else algo
        pol = _q_family(agent_num, action_dim, policy_type)
        if algo == 'vdn':
            pol['model']['mixer'] = False
        pipeline = 'serial'
    elif algo == 'mappo':
        pol, policy_type, pipeline = _mappo(agent_num, action_dim), 'ppo', 'onpolicy'
    elif algo == 'masac':
        pol, policy_type, pipeline = _masac(agent_num, action_dim), 'discrete_sac', 'serial'
    else:
        raise KeyError(f"unknown smac algo: {algo}")
    main_config = EasyDict(dict(
        exp_name=f'smac_{map_name}_{algo}_seed0',
        env=_env_block(map_name),
        policy=pol,
    ))
    create_config = EasyDict(dict(
        env=dict(type='coop_matrix', import_names=['dizoo.multiagent.envs.coop_matrix_env']),
        env_manager=dict(type='base'),
        policy=dict(type=policy_type),
    ))
    main_config._pipeline = pipeline
    return main_config, create_config
