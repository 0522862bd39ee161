from .base_policy import Policy, CommandModePolicy, create_policy, get_policy_cls
from .common_utils import default_preprocess_learn, single_env_forward_wrapper, single_env_forward_wrapper_ttorch
from .dqn import DQNPolicy, DQNSTDIMPolicy
from .ppo import PPOPolicy, PPOPGPolicy, PPOOffPolicy
from .c51 import C51Policy, QRDQNPolicy, IQNPolicy, FQFPolicy, RainbowDQNPolicy, SQLPolicy, MDQNPolicy, BDQPolicy, SQNPolicy
from .ddpg import DDPGPolicy, TD3Policy
from .sac import SACPolicy, DiscreteSACPolicy, SQILSACPolicy
from .a2c import A2CPolicy
from .impala import IMPALAPolicy, PGPolicy
from .qmix import QMIXPolicy, WQMIXPolicy, COMAPolicy
from .offline import BehaviourCloningPolicy, CQLPolicy, DiscreteCQLPolicy, TD3BCPolicy, IQLPolicy, EDACPolicy
from .r2d2 import R2D2Policy
from .dt import DTPolicy
from .ppg import PPGPolicy, PPGOffPolicy
from .acer import ACERPolicy
from .dqfd import DQFDPolicy, PDQNPolicy, D4PGPolicy
from .qmix import MADQNPolicy, CollaQPolicy, QTranPolicy
QTRANPolicy = QTranPolicy  # reference spelling
from .policy_factory import PolicyFactory, get_random_policy
from .r2d2_variants import NGUPolicy, R2D3Policy, R2D2GTrXLPolicy, R2D2CollectTrajPolicy
from .misc_policies import ILPolicy, IBCPolicy, BCQPolicy, TD3VAEPolicy, PromptPGPolicy, PromptAWRPolicy, ProcedureCloningBFSPolicy
from .happo import HAPPOPolicy
from . import command_mode_policy_instance  # registers '<name>_command' variants
from .mbpolicy import MBSACPolicy, STEVESACPolicy
from .dreamer import DREAMERPolicy
from .plan_diffuser import PDPolicy
from .qgpo import QGPOPolicy
from .atoc import ATOCPolicy
from .ppof import PPOFPolicy
# late-registered policies (dreamer/mbpolicy) also need '_command' variants
from .command_mode_policy_instance import _register_command_variants as _rcv
_rcv()
