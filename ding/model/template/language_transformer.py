"""Language-candidate scoring model for prompt policies.

Parity: reference ding/model/template/language_transformer.py
('language_transformer'): encodes a question + N candidate shots and scores
each candidate. Offline note: no pretrained LM weights are downloadable, so
the default text encoder is a hash-embedding bag; a HuggingFace model can be
injected via ``text_encoder`` when transformers weights are available.
"""
import hashlib
from typing import Dict, List, Optional, Union

import torch
import torch.nn as nn

from ding.utils import MODEL_REGISTRY


class HashingTextEncoder(nn.Module):
    """Tokenizer-free text encoder: hashed bag-of-words embedding."""

    def __init__(self, dim: int = 128, buckets: int = 4096):
        super().__init__()
        self.buckets = buckets
        self.embed = nn.EmbeddingBag(buckets, dim, mode='mean')

    def _ids(self, text: str) -> torch.Tensor:
        tokens = text.lower().split()
        if not tokens:
            tokens = ['<empty>']
        ids = [int(hashlib.md5(t.encode()).hexdigest(), 16) % self.buckets for t in tokens]
        return torch.tensor(ids, dtype=torch.long)

    def forward(self, texts: List[str]) -> torch.Tensor:
        device = self.embed.weight.device
        flat, offsets = [], [0]
        for t in texts:
            ids = self._ids(t)
            flat.append(ids)
            offsets.append(offsets[-1] + len(ids))
        flat = torch.cat(flat).to(device)
        offsets = torch.tensor(offsets[:-1], dtype=torch.long, device=device)
        return self.embed(flat, offsets)


@MODEL_REGISTRY.register('language_transformer')
class LanguageTransformer(nn.Module):
    """obs: {'train_sample' (question str), 'candidate_samples' (list[str])}
    -> {'logit': [1, N]} candidate scores."""

    mode = ['compute_actor']

    def __init__(self, model_name: str = 'hash', embedding_size: int = 128, freeze_encoder: bool = False,
                 text_encoder: Optional[nn.Module] = None, **kwargs):
        super().__init__()
        self.encoder = text_encoder or HashingTextEncoder(embedding_size)
        if freeze_encoder:
            for p in self.encoder.parameters():
                p.requires_grad = False
        self.score = nn.Bilinear(embedding_size, embedding_size, 1)

    def forward(self, obs: Dict, mode: str = 'compute_actor') -> Dict:
        question = obs['train_sample'] if isinstance(obs, dict) else str(obs)
        candidates = obs['candidate_samples'] if isinstance(obs, dict) else []
        if isinstance(question, (list, tuple)):
            question = question[0]
        q_emb = self.encoder([question])  # [1, E]
        c_emb = self.encoder(list(candidates))  # [N, E]
        logit = self.score(q_emb.expand(c_emb.shape[0], -1), c_emb).reshape(1, -1)
        return {'logit': logit}
