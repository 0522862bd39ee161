from .network import *  # noqa
from .data_helper import (
    to_device, to_dtype, to_tensor, to_ndarray, to_list, to_item, same_shape, get_tensor_data, zeros_like,
    CudaFetcher, unsqueeze, squeeze, tensor_to_list, get_null_data, build_log_buffer,
)
from .optimizer_helper import (
    Adam, RMSprop, PCGrad, calculate_grad_norm, calculate_grad_norm_without_bias_two_norm, grad_ignore_norm,
    grad_ignore_value, configure_weight_decay,
)
from .checkpoint_helper import build_checkpoint_helper, CheckpointHelper, auto_checkpoint, CountVar
from .loss import LabelSmoothCELoss, SoftLogitsLoss, MultiLogitsLoss, ContrastiveLoss, build_ce_criterion
from .math_helper import cov, unsqueeze_repeat
from .reshape_helper import fold_batch, unfold_batch
from .lr_scheduler import cos_lr_scheduler, get_lr_ratio
from .extras import enable_tf32, get_num_params, levenshtein_distance, hamming_distance, is_differentiable, NonegativeParameter, TanhParameter, CategoricalPd, CategoricalPdPytorch, DataParallel
from ding.utils.default_helper import get_shape0
