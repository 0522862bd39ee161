from .nn_module import (
    MLP, fc_block, conv1d_block, conv2d_block, deconv2d_block, one_hot, binary_encode, NoisyLinearLayer, noise_block,
    build_activation, build_normalization, sequential_pack, normed_linear, normed_conv2d, Lambda, weight_init_,
)
from .res_block import ResBlock, ResFCBlock
from .rnn import LSTM, PytorchLSTM, GRU, get_lstm, sequence_mask, LSTMForwardWrapper
from .transformer import Transformer, TransformerLayer, Attention, ScaledDotProductAttention
from .gtrxl import GTrXL, GRUGatingUnit, PositionalEmbedding, AttentionXL, Memory
from .scatter_connection import ScatterConnection
from .popart import PopArt
from .blocks_extra import (
    Swish, Flatten, NearestUpsample, BilinearUpsample, SoftArgmax, GumbelSoftmax, GatingType, SumMerge, VectorMerge,
    ResNet, resnet18,
)
from .nn_module import NoisyLinearLayer as NoiseLinearLayer  # reference spelling
