"""One-off MI355X breadth sweep: force cuda=True on a batch of zoo configs
beyond the pytest GPU subset and run each for one collect->train iteration.
Usage: python ding/scripts/gpu_zoo_sweep.py [n]  (prints OK/FAIL per config).
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), '..', '..')))

CASES = [
    ('dizoo.atari.config.serial.spaceinvaders_fqf_config', 'serial'),
    ('dizoo.atari.config.serial.qbert_iqn_config', 'serial'),
    ('dizoo.atari.config.serial.pong_acer_config', 'serial'),
    ('dizoo.atari.config.serial.qbert_a2c_config', 'onpolicy'),
    ('dizoo.atari.config.serial.pong_sql_config', 'serial'),
    ('dizoo.atari.config.serial.enduro_impala_config', 'serial'),
    ('dizoo.atari.config.serial.pong_stdim_config', 'serial'),
    ('dizoo.smac.config.smac_MMM_qtran_config', 'serial'),
    ('dizoo.smac.config.smac_25m_mappo_config', 'onpolicy'),
    ('dizoo.smac.config.smac_3s5z_collaq_config', 'serial'),
    ('dizoo.mujoco.config.walker2d_td3_config', 'serial'),
    ('dizoo.mujoco.config.humanoid_onppo_config', 'onpolicy'),
    ('dizoo.mujoco.config.halfcheetah_bdq_config', 'serial'),
    ('dizoo.box2d.lunarlander.config.lunarlander_rainbow_config', 'serial'),
    ('dizoo.box2d.lunarlander.config.lunarlander_r2d2_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_sac_config', 'serial'),
]


def main():
    import tests.test_dizoo_smoke as smoke
    from tests.test_dizoo_smoke import _run_one

    orig = smoke._shrink

    def cuda_shrink(m, c):
        m2, c2 = orig(m, c)
        m2.policy.cuda = True
        return m2, c2

    smoke._shrink = cuda_shrink
    limit = int(sys.argv[1]) if len(sys.argv) > 1 else len(CASES)
    fails = 0
    for mod, pipe in CASES[:limit]:
        try:
            with tempfile.TemporaryDirectory() as d:
                _run_one(mod, pipe, tmp_dir=d)
            print('OK  ', mod, flush=True)
        except Exception as e:
            fails += 1
            print('FAIL', mod, type(e).__name__, str(e)[:160], flush=True)
    print(f'SWEEP_DONE fails={fails}/{limit}')


if __name__ == '__main__':
    main()
