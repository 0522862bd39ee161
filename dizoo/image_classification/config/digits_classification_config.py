"""Synthetic-digits supervised classification (reference
dizoo/image_classification/entry/imagenet_res18_config.py shape)."""
from ding.utils import EasyDict

digits_classification_config = EasyDict(dict(
    exp_name='digits_classification_seed0',
    env=dict(),
    policy=dict(
        cuda=True,
        model=dict(obs_shape=[1, 28, 28], action_shape=10, encoder_hidden_size_list=[32, 64, 128]),
        learn=dict(batch_size=64, learning_rate=0.01, weight_decay=1e-4, update_per_collect=1),
        collect=dict(unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=100, )),
    ),
))
main_config = digits_classification_config
digits_classification_create_config = EasyDict(dict(
    policy=dict(type='image_classification', import_names=['dizoo.image_classification.policy']),
))
create_config = digits_classification_create_config
