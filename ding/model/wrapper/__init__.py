from .model_wrappers import model_wrap, IModelWrapper, wrapper_name_map, register_wrapper, TargetNetworkWrapper, HiddenStateWrapper
