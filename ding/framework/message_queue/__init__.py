from .mq import MQ
from .tcp import TCPMQ
from .nng import NNGMQ
from .redis import RedisMQ
