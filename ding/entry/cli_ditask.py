"""``ditask`` CLI: launch a task main() across parallel workers on the event
bus.

Parity: reference ding/entry/cli_ditask.py:66.
"""
import importlib
import os
import sys

import click

from ding import __VERSION__


@click.command(context_settings=dict(help_option_names=['-h', '--help']))
@click.version_option(version=__VERSION__)
@click.option('-m', '--main', type=str, required=True, help='module path of main, e.g. pkg.mod.main')
@click.option('--parallel-workers', type=int, default=1, help='number of local workers')
@click.option('--topology', type=click.Choice(['mesh', 'star', 'alone']), default='mesh')
@click.option('--protocol', type=click.Choice(['tcp']), default='tcp')
@click.option('--address', type=str, default='127.0.0.1')
@click.option('--ports', type=int, default=None, help='starting port (auto if omitted)')
@click.option('--attach-to', type=str, default=None, help='comma-separated remote node addrs')
@click.option('--node-ids', type=str, default=None, help='comma-separated node id overrides')
@click.option('--labels', type=str, default=None, help='comma-separated node labels')
@click.option('--mq-type', type=str, default='tcp')
@click.option('--auto-recover', is_flag=True, default=False)
@click.option('--max-retries', type=int, default=1)
def cli_ditask(main, parallel_workers, topology, protocol, address, ports, attach_to, node_ids, labels, mq_type,
               auto_recover, max_retries):
    from ding.framework.parallel import Parallel
    sys.path.insert(0, os.getcwd())
    mod_name, fn_name = main.rsplit('.', 1)
    module = importlib.import_module(mod_name)
    main_fn = getattr(module, fn_name)
    attach_list = attach_to.split(',') if attach_to else []
    node_id_list = [int(x) for x in node_ids.split(',')] if node_ids else None
    label_set = set(labels.split(',')) if labels else None
    Parallel.runner(
        n_parallel_workers=parallel_workers,
        mq_type=mq_type,
        address=address,
        ports=ports,
        topology=topology,
        attach_to=attach_list,
        node_ids=node_id_list,
        labels=label_set,
        auto_recover=auto_recover,
        max_retries=max_retries,
    )(main_fn)


if __name__ == '__main__':
    cli_ditask()
