"""Factory for the Atari per-game config families. The reference ships one
hand-written config file per (game, algorithm) pair
(dizoo/atari/config/serial/<game>/<game>_<algo>_config.py); here one factory
builds the same (main_config, create_config) pairs so every game shares one
tuned parameter block per algorithm. Thin per-name modules (qbert_dqn_config
etc.) call :func:`build_atari_config` so the reference's module paths keep
working.

All configs run on the atari-lite synthetic env (no ALE in this image): same
obs/action/reward interface and compute shape, with a learnable signal.
"""
import copy

from ding.utils import EasyDict

# game -> (env_id, action_num, dqn-family stop value)
GAMES = {
    'pong': ('PongNoFrameskip-v4', 6, 18),
    'qbert': ('QbertNoFrameskip-v4', 6, 30000),
    'spaceinvaders': ('SpaceInvadersNoFrameskip-v4', 6, 2000),
    'enduro': ('EnduroNoFrameskip-v4', 9, 700),
    'asterix': ('AsterixNoFrameskip-v4', 9, 10000),
    'demon_attack': ('DemonAttackNoFrameskip-v4', 6, 8000),
    'phoenix': ('PhoenixNoFrameskip-v4', 8, 10000),
    'pitfall': ('PitfallNoFrameskip-v4', 18, 0),
    'montezuma': ('MontezumaRevengeNoFrameskip-v4', 18, 100),
}

_OBS = [4, 84, 84]
_ENC = [128, 128, 512]


def _env_block(game: str) -> dict:
    env_id, action_num, stop = GAMES[game]
    return dict(
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=stop,
        env_id=env_id,
        action_num=action_num,
        frame_stack=4,
    )


def _q_other() -> dict:
    return dict(
        eps=dict(type='exp', start=1., end=0.05, decay=250000),
        replay_buffer=dict(replay_buffer_size=100000, ),
    )


def _policy_block(algo: str, action_num: int) -> (dict, str, str):
    """Return (policy_cfg, policy_type, pipeline)."""
    model = dict(obs_shape=_OBS, action_shape=action_num, encoder_hidden_size_list=_ENC)
    if algo in ('dqn', 'mdqn', 'sql', 'stdim'):
        pol = dict(
            cuda=True,
            model=copy.deepcopy(model),
            nstep=3,
            discount_factor=0.99,
            learn=dict(update_per_collect=10, batch_size=32, learning_rate=1e-4, target_update_freq=500),
            collect=dict(n_sample=96, ),
            eval=dict(evaluator=dict(eval_freq=4000, )),
            other=_q_other(),
        )
        if algo == 'mdqn':
            pol['nstep'] = 1  # Munchausen target is 1-step
            pol['learn']['entropy_tau'] = 0.03
            pol['learn']['m_alpha'] = 0.9
            return pol, 'mdqn', 'serial'
        if algo == 'sql':
            pol['learn']['alpha'] = 0.12
            return pol, 'sql', 'serial'
        if algo == 'stdim':
            pol['aux_model'] = dict(encode_shape=64, heads=[1, 1], loss_type='infonce')
            pol['aux_loss_weight'] = 0.003
            return pol, 'dqn_stdim', 'serial'
        return pol, 'dqn', 'serial'
    if algo in ('c51', 'rainbow'):
        m = dict(**copy.deepcopy(model), v_min=-10, v_max=10, n_atom=51)
        pol = dict(
            cuda=True,
            priority=(algo == 'rainbow'),
            model=m,
            nstep=3,
            discount_factor=0.99,
            learn=dict(update_per_collect=10, batch_size=32, learning_rate=1e-4, target_update_freq=500),
            collect=dict(n_sample=96, ),
            eval=dict(evaluator=dict(eval_freq=4000, )),
            other=_q_other(),
        )
        return pol, algo, 'serial'
    if algo in ('qrdqn', 'iqn', 'fqf'):
        m = copy.deepcopy(model)
        m['num_quantiles'] = 32
        if algo in ('iqn', 'fqf'):
            m['quantile_embedding_size'] = 64
        pol = dict(
            cuda=True,
            priority=False,
            model=m,
            nstep=3,
            discount_factor=0.99,
            learn=dict(
                update_per_collect=10,
                batch_size=32,
                learning_rate=1e-4 if algo != 'fqf' else 5e-5,
                target_update_freq=500,
                kappa=1.0,
            ),
            collect=dict(n_sample=96, ),
            eval=dict(evaluator=dict(eval_freq=4000, )),
            other=_q_other(),
        )
        return pol, algo, 'serial'
    if algo in ('a2c', 'onppo'):
        enc = [64, 64, 128]
        pol = dict(
            cuda=True,
            action_space='discrete',
            model=dict(
                obs_shape=_OBS,
                action_shape=action_num,
                action_space='discrete',
                encoder_hidden_size_list=enc,
                actor_head_hidden_size=enc[-1],
                critic_head_hidden_size=enc[-1],
            ),
            learn=dict(
                epoch_per_collect=10,
                update_per_collect=1,
                batch_size=320,
                learning_rate=3e-4,
                value_weight=0.5,
                entropy_weight=0.001,
                adv_norm=True,
            ),
            collect=dict(n_sample=3200, discount_factor=0.99, gae_lambda=0.95),
            eval=dict(evaluator=dict(eval_freq=1000, )),
        )
        if algo == 'a2c':
            pol['learn'].pop('epoch_per_collect')
            pol['learn']['batch_size'] = 160
            pol['collect']['n_sample'] = 160
            return pol, 'a2c', 'onpolicy'
        pol['recompute_adv'] = True
        return pol, 'ppo', 'onpolicy'
    if algo == 'offppo':
        enc = [64, 64, 128]
        pol = dict(
            cuda=True,
            model=dict(
                obs_shape=_OBS,
                action_shape=action_num,
                action_space='discrete',
                encoder_hidden_size_list=enc,
                actor_head_hidden_size=enc[-1],
                critic_head_hidden_size=enc[-1],
            ),
            learn=dict(update_per_collect=24, batch_size=128, learning_rate=1e-4, entropy_weight=0.01),
            collect=dict(n_sample=1024, discount_factor=0.99, gae_lambda=0.95),
            eval=dict(evaluator=dict(eval_freq=1000, )),
            other=dict(replay_buffer=dict(replay_buffer_size=10000, )),
        )
        return pol, 'ppo_offpolicy', 'serial'
    if algo == 'ppg':
        enc = [64, 64, 128]
        pol = dict(
            cuda=True,
            action_space='discrete',
            model=dict(
                obs_shape=_OBS,
                action_shape=action_num,
                action_space='discrete',
                encoder_hidden_size_list=enc,
                actor_head_hidden_size=enc[-1],
                critic_head_hidden_size=enc[-1],
            ),
            learn=dict(
                epoch_per_collect=10,
                batch_size=320,
                learning_rate=3e-4,
                value_weight=0.5,
                entropy_weight=0.001,
                epochs_aux=6,
                beta_weight=1.0,
                aux_freq=1,
            ),
            collect=dict(n_sample=3200, discount_factor=0.99, gae_lambda=0.95),
            eval=dict(evaluator=dict(eval_freq=1000, )),
        )
        return pol, 'ppg', 'onpolicy_ppg'
    if algo == 'acer':
        pol = dict(
            cuda=True,
            model=dict(obs_shape=_OBS, action_shape=action_num, encoder_hidden_size_list=[64, 64, 128]),
            learn=dict(update_per_collect=4, batch_size=16, learning_rate=3e-4, c_clip_ratio=10, trust_region=True),
            collect=dict(n_sample=64, ),
            eval=dict(evaluator=dict(eval_freq=1000, )),
            other=dict(replay_buffer=dict(replay_buffer_size=5000)),
        )
        return pol, 'acer', 'serial'
    if algo == 'impala':
        pol = dict(
            cuda=True,
            model=dict(
                obs_shape=_OBS,
                action_shape=action_num,
                encoder_hidden_size_list=[128, 128, 256],
                critic_head_hidden_size=256,
                actor_head_hidden_size=256,
            ),
            unroll_len=32,
            learn=dict(
                update_per_collect=2,
                batch_size=128,
                learning_rate=6e-4,
                value_weight=0.5,
                entropy_weight=0.01,
                discount_factor=0.99,
                lambda_=0.95,
                rho_clip_ratio=1.0,
                c_clip_ratio=1.0,
            ),
            collect=dict(n_sample=16, ),
            eval=dict(evaluator=dict(eval_freq=1000, )),
            other=dict(replay_buffer=dict(replay_buffer_size=1000, sliced=True)),
        )
        return pol, 'impala', 'serial'
    if algo in ('r2d2', 'r2d2_gtrxl'):
        pol = dict(
            cuda=True,
            priority=True,
            priority_IS_weight=True,
            model=dict(
                obs_shape=_OBS,
                action_shape=action_num,
                encoder_hidden_size_list=[128, 128, 512],
            ),
            discount_factor=0.997,
            nstep=5,
            burnin_step=2,
            unroll_len=40,
            learn_unroll_len=38,
            learn=dict(update_per_collect=8, batch_size=64, learning_rate=5e-4, target_update_theta=0.001),
            collect=dict(n_sample=32, unroll_len=40, env_num=8),
            eval=dict(evaluator=dict(eval_freq=4000, ), env_num=8),
            other=dict(
                eps=dict(type='exp', start=0.95, end=0.05, decay=100000),
                replay_buffer=dict(replay_buffer_size=10000, ),
            ),
        )
        if algo == 'r2d2_gtrxl':
            pol['model'].update(memory_len=8, att_head_num=4, att_head_dim=16, hidden_size=64, att_layer_num=2)
            pol['learn_unroll_len'] = 20
            pol['unroll_len'] = 20
            pol['collect']['unroll_len'] = 20
            pol.pop('burnin_step')
            return pol, 'r2d2_gtrxl', 'serial'
        pol['model']['lstm_type'] = 'normal'
        return pol, 'r2d2', 'serial'
    raise KeyError(f"unknown atari algo: {algo}")


def build_atari_config(game: str, algo: str) -> (EasyDict, EasyDict):
    """(main_config, create_config) for one (game, algorithm) pair."""
    env_id, action_num, _ = GAMES[game]
    policy, policy_type, pipeline = _policy_block(algo, action_num)
    main_config = EasyDict(dict(
        exp_name=f'{game}_{algo}_seed0',
        env=_env_block(game),
        policy=policy,
    ))
    create_config = EasyDict(dict(
        env=dict(type='atari_lite', import_names=['dizoo.atari.envs.atari_lite_env']),
        env_manager=dict(type='subprocess'),
        policy=dict(type=policy_type),
    ))
    if pipeline == 'serial' and policy_type in ('impala', ):
        create_config.replay_buffer = dict(type='naive')
    main_config._pipeline = pipeline
    return main_config, create_config
