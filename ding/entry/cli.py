"""``ding`` CLI.

Parity: reference ding/entry/cli.py:135 (modes serial/parallel/dist/eval +
variants).
"""
import os

import click

from ding import __VERSION__

CONTEXT_SETTINGS = dict(help_option_names=['-h', '--help'])


@click.command(context_settings=CONTEXT_SETTINGS)
@click.version_option(version=__VERSION__)
@click.option('-m', '--mode', type=click.Choice([
    'serial', 'serial_onpolicy', 'serial_offline', 'serial_reward_model', 'serial_gail', 'serial_sqil',
    'serial_dqfd', 'parallel', 'dist', 'eval',
]), default='serial', help='run mode')
@click.option('-c', '--config', type=str, help='path to the config .py/.yaml file')
@click.option('-s', '--seed', type=int, default=0, help='random seed')
@click.option('--env', type=str, default=None, help='env shortcut (with -p policy)')
@click.option('-p', '--policy', type=str, default=None, help='policy shortcut (with --env)')
@click.option('--train-iter', type=int, default=int(1e10), help='max train iterations')
@click.option('--env-step', type=int, default=int(1e10), help='max env steps')
@click.option('--load-path', type=str, default=None, help='checkpoint to load (eval mode)')
@click.option('--replay-path', type=str, default=None, help='replay save dir (eval mode)')
def cli(mode, config, seed, env, policy, train_iter, env_step, load_path, replay_path):
    if config is None and (env is None or policy is None):
        raise click.UsageError('provide -c CONFIG or --env with -p POLICY')
    if mode == 'serial':
        from .serial_entry import serial_pipeline
        serial_pipeline(config, seed, max_train_iter=train_iter, max_env_step=env_step)
    elif mode == 'serial_onpolicy':
        from .serial_entry import serial_pipeline_onpolicy
        serial_pipeline_onpolicy(config, seed, max_train_iter=train_iter, max_env_step=env_step)
    elif mode == 'serial_offline':
        from .serial_entry_offline import serial_pipeline_offline
        serial_pipeline_offline(config, seed, max_train_iter=train_iter)
    elif mode == 'serial_reward_model':
        from .serial_entry_variants import serial_pipeline_reward_model
        serial_pipeline_reward_model(config, seed, max_train_iter=train_iter, max_env_step=env_step)
    elif mode == 'eval':
        from .application_entry import eval as eval_entry
        value = eval_entry(config, seed, load_path=load_path, replay_path=replay_path)
        click.echo(f'Eval episode return: {value:.3f}')
    elif mode == 'parallel':
        from .parallel_entry import parallel_pipeline
        parallel_pipeline(config, seed)
    elif mode == 'dist':
        # legacy role launcher: --platform slurm/k8s fills per-process args;
        # the modern multi-process path is `ditask`
        from ding.entry.dist_entry import dist_prepare_config
        click.echo(
            'dist mode (legacy): use ding.entry.dist_entry launchers '
            '(dist_launch_coordinator/learner/collector) or `ditask` for the event-bus runtime'
        )
        return
    else:
        raise click.UsageError(f'unimplemented mode {mode}')


if __name__ == '__main__':
    cli()
