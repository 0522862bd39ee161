from .q_learning import DQN, BDQ, C51DQN, QRDQN, IQN, FQF, RainbowDQN, DRQN, GTrXLDQN
from .vac import VAC, DREAMERVAC
from .qac import ContinuousQAC, DiscreteQAC
from .pg import PG
from .qmix import QMix, Mixer
from .coma_model import COMA
from .mavac import MAVAC
from .bc import DiscreteBC, ContinuousBC, EDAC
from .decision_transformer import DecisionTransformer
from .acer_model import ACER
from .pdqn import PDQN
from .qac_dist import QACDIST
from .ebm import EBM, DFO, LangevinMCMC
from .vae import VanillaVAE
from .language_transformer import LanguageTransformer
from .hpt import HPT, PolicyStem
from .diffusion import PlanDiffuser, GaussianDiffusion, ValueDiffusion, GaussianInvDynDiffusion
from .qgpo import QGPO
from .atoc import ATOC
from .marl_models import MADQN, WQMix, QTran, CollaQ, HAVAC
from .maqac import DiscreteMAQAC, ContinuousMAQAC
from .extras import SQN, PPG, BCQ, NGU, ProcedureCloningBFS, ProcedureCloningMCTS, AutoregressiveEBM, ContinuousQVAC
