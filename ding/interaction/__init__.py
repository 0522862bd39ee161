from .master import Master, Slave, HttpEngine, success_response, failure_response
