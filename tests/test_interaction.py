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
