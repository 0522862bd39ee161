from .common import *  # noqa
from .template import *  # noqa
from .wrapper import model_wrap, IModelWrapper, TargetNetworkWrapper, HiddenStateWrapper
