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
