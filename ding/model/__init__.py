from .common import *  # noqa
from .template import *  # noqa
from .wrapper import model_wrap, IModelWrapper, TargetNetworkWrapper, HiddenStateWrapper, \
    wrapper_name_map, register_wrapper
