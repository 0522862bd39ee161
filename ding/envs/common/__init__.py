from .spaces import Space, Discrete, Box, MultiDiscrete, Dict
from .common_function import (one_hot, sqrt_one_hot, div_one_hot, div_func, clip_one_hot,
                              batch_binary_encode, get_postion_vector, affine_transform)
