"""Representative dizoo configs running on the MI355X with cuda=True —
one collect->train iteration each through the real pipelines (the CPU
smoke sweep covers all ~115 configs; this proves the zoo's policies run
on-device: conv stacks, recurrent unrolls, MARL mixers, offline learners).
"""
import copy
import importlib

import pytest

from tests.test_dizoo_smoke import SMOKE_CONFIGS, _run_one, _shrink  # noqa

pytestmark = pytest.mark.gpu

GPU_SUBSET = [
    ('dizoo.atari.config.serial.pong_dqn_config', 'serial'),
    ('dizoo.atari.config.serial.pong_ppo_config', 'onpolicy'),
    ('dizoo.mujoco.config.hopper_sac_config', 'serial'),
    ('dizoo.procgen.config.coinrun_ppo_config', 'onpolicy'),
    ('dizoo.minigrid.config.minigrid_r2d2_config', 'serial'),
    ('dizoo.petting_zoo.config.ptz_simple_spread_qmix_config', 'serial'),
    ('dizoo.d4rl.config.hopper_medium_td3bc_config', 'offline'),
    ('dizoo.box2d.lunarlander.config.lunarlander_dqn_config', 'serial'),
    ('dizoo.dmc2gym.config.dmc2gym_sac_pixel_config', 'serial'),
    ('dizoo.mario.config.mario_dqn_config', 'serial'),
    # breadth extension (round 2): distinct on-device codepaths — hybrid
    # action PDQN, factor-sequential HAPPO, distributional d4pg, GTrXL
    # attention unrolls, quantile fqf, offline decision transformer,
    # recurrent NGU intrinsic stack, dreamer RSSM, multi-agent ATOC comm
    ('dizoo.gym_hybrid.config.gym_hybrid_pdqn_config', 'serial'),
    ('dizoo.petting_zoo.config.ptz_simple_spread_happo_config', 'onpolicy'),
    ('dizoo.classic_control.pendulum.config.pendulum_d4pg_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_r2d2_gtrxl_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_fqf_config', 'serial'),
    ('dizoo.d4rl.config.hopper_expert_dt_config', 'offline_dt'),
    ('dizoo.dmc2gym.config.cartpole_balance_dreamer_config', 'dreamer'),
    ('dizoo.petting_zoo.config.ptz_simple_spread_atoc_config', 'serial'),
    # round-2 late families: factory-built atari/smac/mujoco configs
    ('dizoo.atari.config.serial.qbert_rainbow_config', 'serial'),
    ('dizoo.atari.config.serial.pong_mdqn_config', 'serial'),
    ('dizoo.smac.config.smac_3s5z_masac_config', 'serial'),
    ('dizoo.mujoco.config.hopper_d4pg_config', 'serial'),
]


def _force_cuda(module_name: str, pipeline: str, tmp_path):
    mod = importlib.import_module(module_name)
    # _run_one shrinks and runs; patch cuda back on through a wrapper module
    import tests.test_dizoo_smoke as smoke

    orig = smoke._shrink

    def cuda_shrink(m, c):
        m2, c2 = orig(m, c)
        m2.policy.cuda = True
        return m2, c2

    smoke._shrink = cuda_shrink
    try:
        _run_one(module_name, pipeline, tmp_dir=str(tmp_path))
    finally:
        smoke._shrink = orig


@pytest.mark.parametrize('module_name,pipeline', GPU_SUBSET)
def test_dizoo_config_on_gpu(module_name, pipeline, tmp_path):
    _force_cuda(module_name, pipeline, tmp_path)
