from .master import Master, Slave, HttpEngine, TaskFail, success_response, failure_response
