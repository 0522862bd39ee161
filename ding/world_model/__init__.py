from .base_world_model import (
    WorldModel, DynaWorldModel, DreamWorldModel, HybridWorldModel, create_world_model, get_world_model_cls,
    get_rollout_length_scheduler,
)
from .mbpo import MBPOWorldModel, EnsembleModel, EnsembleFC
from .ddppo import DDPPOWorldMode
from .dreamer import DREAMERWorldModel, RSSM, ConvDecoder
from .idm import InverseDynamicsModel
