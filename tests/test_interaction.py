"""HTTP master/slave round-trip."""
import time

import pytest

from ding.interaction import Master, Slave


class EchoSlave(Slave):

    def _process_task(self, task):
        return {'echo': task.get('x', 0) * 2}


def test_master_slave_roundtrip():
    slave = EchoSlave().start()
    master = Master().start()
    try:
        master.connect_slave('s0', '127.0.0.1', slave.port)
        task_id = master.new_task('s0', {'x': 21})
        result = master.wait_task(task_id, timeout=10)
        assert result['status'] == 'done'
        assert result['result']['echo'] == 42
    finally:
        master.close()
        slave.close()


def test_coordinator_dispatch():
    from ding.worker import Coordinator
    slave = EchoSlave().start()
    coord = Coordinator().start()
    try:
        coord.register_worker('w0', '127.0.0.1', slave.port)
        tid = coord.submit_task({'x': 5})
        result = coord.wait_task(tid, timeout=15)
        assert result['result']['echo'] == 10
    finally:
        coord.close()
        slave.close()


def test_learner_aggregator():
    from ding.worker import LearnerAggregator
    agg = LearnerAggregator([lambda: {'loss': 1.0}, lambda: {'loss': 3.0}])
    info = agg.merge_info()
    assert info['loss'] == 2.0 and info['learner_num'] == 2

    def dead():
        raise ConnectionError('down')

    agg2 = LearnerAggregator([lambda: {'loss': 1.0}, dead])
    info2 = agg2.merge_info()
    assert info2['dead_learner_num'] == 1


def test_comm_learner_fs_roundtrip(tmp_path):
    """FlaskFileSystemLearner: start -> learn from FS data -> policy saved."""
    import torch
    from ding.worker.comm import FlaskFileSystemLearner
    from ding.utils import EasyDict
    from tests.test_policy_breadth import cartpole_cfg
    from ding.config import compile_config

    main, create = cartpole_cfg('dqn', extra_policy=dict(nstep=1))
    create.policy.type = 'dqn_command'
    cfg = compile_config(main, create_cfg=create, auto=True, save_cfg=False)

    learner_worker = FlaskFileSystemLearner(
        EasyDict({'path_data': str(tmp_path / 'data'), 'path_policy': str(tmp_path / 'policy')})
    )
    out = learner_worker._process_task({'name': 'resource'})
    assert 'cpu' in out
    out = learner_worker._process_task({
        'name': 'learner_start_task',
        'task_info': {
            'policy': dict(cfg.policy, type='dqn'), 'learner_cfg': {}, 'policy_id': 'p.pth',
            'exp_name': str(tmp_path / 'exp'),
        },
    })
    assert 'started' in out['message']
    demand = learner_worker._process_task({'name': 'learner_get_data_task', 'task_id': 't0', 'buffer_id': 'b0'})
    assert demand['batch_size'] == cfg.policy.learn.batch_size
    # fabricate transitions on the FS
    data = [
        {
            'obs': torch.randn(4), 'next_obs': torch.randn(4), 'action': torch.tensor([0]),
            'reward': torch.randn(1), 'done': False, 'collect_iter': 0,
        } for _ in range(demand['batch_size'])
    ]
    torch.save(data, tmp_path / 'data' / 'batch0.pth')
    out = learner_worker._process_task({'name': 'learner_learn_task', 'data': ['batch0.pth']})
    assert out['train_iter'] >= 1
    assert (tmp_path / 'policy' / 'p.pth').exists()
    learner_worker._process_task({'name': 'learner_close_task'})


def test_comm_collector_fs_roundtrip(tmp_path):
    import torch
    from ding.worker.comm import FlaskFileSystemCollector
    from ding.utils import EasyDict
    from tests.test_policy_breadth import cartpole_cfg
    from ding.config import compile_config

    main, create = cartpole_cfg('dqn')
    create.policy.type = 'dqn_command'
    cfg = compile_config(main, create_cfg=create, auto=True, save_cfg=False)
    cfg.policy.type = 'dqn'

    worker = FlaskFileSystemCollector(EasyDict({'path_data': str(tmp_path), 'path_policy': str(tmp_path)}))
    out = worker._process_task({'name': 'collector_start_task', 'task_info': {'cfg': cfg}})
    assert 'started' in out['message']
    out = worker._process_task({'name': 'collector_data_task', 'n_sample': 8,
                                'policy_kwargs': {'eps': 0.5}})
    assert out['sample_count'] == 8
    data = torch.load(tmp_path / out['data_path'], weights_only=False)
    assert len(data) == 8 and 'obs' in data[0]
    worker._process_task({'name': 'collector_close_task'})


def test_dist_entry_in_process(tmp_path):
    """Legacy distributed loop: commander -> collector task -> FS data ->
    learner task -> policy saved -> next collector cycle reloads it."""
    from ding.entry.dist_entry import dist_launch_coordinator
    from ding.worker.comm import FlaskFileSystemCollector, FlaskFileSystemLearner
    from ding.utils import EasyDict
    from ding.config import compile_config
    from tests.test_policy_breadth import cartpole_cfg

    main, create = cartpole_cfg('dqn', extra_policy=dict(nstep=1))
    create.policy.type = 'dqn_command'
    cfg = compile_config(main, create_cfg=create, auto=True, save_cfg=False)
    cfg.policy.type = 'dqn'

    comm_cfg = EasyDict({'path_data': str(tmp_path), 'path_policy': str(tmp_path)})
    learner = FlaskFileSystemLearner(comm_cfg)
    collector = FlaskFileSystemCollector(comm_cfg)
    coord_cfg = EasyDict(dict(commander=dict(
        type='solo', eval_interval=int(1e9),
        collector_cfg=dict(cfg=cfg, n_sample=16),
        policy=dict(cfg.policy),
        policy_id='p.pth',
    )))
    hist = dist_launch_coordinator(coord_cfg, learner=learner, collector=collector, max_cycles=3)
    assert len(hist['collect']) >= 1
    assert len(hist['learn']) >= 1
    assert hist['learn'][0]['train_iter'] >= 1
