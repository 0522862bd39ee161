from .encoder import (
    ConvEncoder, FCEncoder, IMPALAConvEncoder, StructEncoder, GaussianFourierProjectionTimeEncoder,
    IMPALACnnResidualBlock, IMPALACnnDownStack,
)
from .head import (
    independent_normal_dist,
    DiscreteHead, DistributionHead, RainbowHead, QRDQNHead, QuantileHead, FQFHead, DuelingHead, BranchingHead,
    StochasticDuelingHead, RegressionHead, ReparameterizationHead, PopArtVHead, AttentionPolicyHead, MultiHead,
    EnsembleHead, head_cls_map,
)
from .utils import create_model, top_p_logits
