from .functional import *  # noqa
from .collector import PPOFStepCollector, StepCollector, EpisodeCollector
from .learner import OffPolicyLearner, HERLearner
from .ckpt_handler import CkptSaver
from .distributer import ContextExchanger, ModelExchanger, PeriodicalModelExchanger
from .barrier import Barrier, BarrierRuntime
from .gpu_exchanger import gpu_trajectory_sender, gpu_trajectory_receiver
