"""TabMWP prompt_awr (reference dizoo/tabmwp/config/tabmwp_prompt_awr_config.py):
prompt-selection RL with the hash-embedding language model."""
from ding.utils import EasyDict

tabmwp_prompt_awr_config = EasyDict(dict(
    exp_name='tabmwp_prompt_awr_seed0',
    env=dict(
        collector_env_num=1,
        evaluator_env_num=1,
        n_evaluator_episode=20,
        cand_number=4,
        stop_value=1,
    ),
    policy=dict(
        cuda=True,
        shot_number=1,
        model=dict(embedding_size=128),
        learn=dict(batch_size=16, learning_rate=1e-4, entropy_weight=0.001, grad_norm=0.5),
        collect=dict(unroll_len=1, discount_factor=1.0, n_episode=16,
                     collector=dict(get_train_sample=True, type='episode')),
        eval=dict(evaluator=dict(eval_freq=50, )),
    ),
))
main_config = tabmwp_prompt_awr_config
tabmwp_prompt_awr_create_config = EasyDict(dict(
    env=dict(type='tabmwp', import_names=['dizoo.tabmwp.envs.tabmwp_env']),
    env_manager=dict(type='base'),
    policy=dict(type='prompt_awr'),
))
create_config = tabmwp_prompt_awr_create_config
