"""Model factory. Parity: reference ding/model/common/utils.py:7 create_model."""
import torch

from ding.utils import MODEL_REGISTRY, EasyDict


def create_model(cfg: EasyDict) -> torch.nn.Module:
    """Build a registered model from cfg; ``cfg.type`` selects the template,
    remaining keys are constructor kwargs."""
    cfg = EasyDict(cfg)
    import_names = cfg.pop('import_names', [])
    for name in import_names:
        __import__(name)
    model_type = cfg.pop('type')
    return MODEL_REGISTRY.build(model_type, **cfg)


def top_p_logits(logits: torch.Tensor, topp: float = 0.9, filter_value: float = 0, min_topk: int = 1):
    """Nucleus filtering over the last dim (language-policy sampling)."""
    cum_logits = logits.clone()
    if topp > 0:
        probs = torch.softmax(logits, dim=-1)
        sorted_probs, sorted_idx = probs.sort(dim=-1, descending=True)
        cumsum = sorted_probs.cumsum(dim=-1)
        mask = cumsum - sorted_probs > topp
        mask[..., :min_topk] = False
        remove_mask = torch.zeros_like(mask).scatter_(-1, sorted_idx, mask)
        cum_logits[remove_mask] = filter_value
        cum_logits.div_(cum_logits.sum(dim=-1, keepdim=True).clamp(min=1e-8))
    return cum_logits
