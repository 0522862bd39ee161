from .base_reward_model import BaseRewardModel, create_reward_model, get_reward_model_cls
from .exploration import RndRewardModel, ICMRewardModel
from .imitation import (
    GailRewardModel, GuidedCostRewardModel, PwilRewardModel, RedRewardModel, PdeilRewardModel, TrexRewardModel,
    DrexRewardModel,
)
from .her_reward_model import HerRewardModel
from .ngu_reward_model import RndNGURewardModel, EpisodicNGURewardModel
