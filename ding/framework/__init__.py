from .context import Context, OnlineRLContext, OfflineRLContext
from .event_loop import EventLoop
from .task import Task, task, Role, VoidMiddleware
from .parallel import Parallel
from .supervisor import Supervisor, ChildType, SendPayload, RecvPayload


def ding_init(cfg):
    """Bind the experiment-wide DistributedWriter (reference
    framework/__init__.py:10)."""
    from ding.utils import DistributedWriter
    return DistributedWriter.get_instance(cfg.exp_name + "/log")
from .wrapper import StepTimer
