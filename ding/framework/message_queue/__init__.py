from .mq import MQ
from .tcp import TCPMQ
