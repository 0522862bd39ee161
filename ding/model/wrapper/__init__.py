from .model_wrappers import model_wrap, IModelWrapper, wrapper_name_map, TargetNetworkWrapper, HiddenStateWrapper
