"""RLHF / LLM datasets: prompt datasets for online RL (PPO-style) and
prompt+response+label datasets for offline RL (DPO/KTO-style).

Parity: reference ding/utils/data/rlhf_online_dataset.py (OnlineRLDataset)
and ding/utils/data/rlhf_offline_dataset.py (OfflineRLDataset,
zero_pad_sequences). The tokenizer is any HuggingFace-style callable
(``tokenizer(text, return_tensors='pt', ...) -> {'input_ids',
'attention_mask'}`` with ``eos_token`` / ``eos_token_id`` attributes) —
offline images have no hub access, so tests drive these with a local stub
tokenizer and production uses transformers.AutoTokenizer.
"""
from typing import Any, Callable, Dict, Iterable, List, Union

import torch
import torch.nn.functional as F
from torch.utils.data import Dataset


def zero_pad_sequences(sequences: List[torch.Tensor], side: str = "left", value: int = 0) -> torch.Tensor:
    """Pad 1-D (or [1, L]) token tensors to a common length and stack to
    [B, L_max]. ``side`` picks which end receives the padding."""
    assert side in ("left", "right"), side
    longest = max(s.size(-1) for s in sequences)
    out = []
    for s in sequences:
        gap = longest - s.size(-1)
        pad = (gap, 0) if side == "left" else (0, gap)
        out.append(F.pad(s, pad, value=value))
    return torch.stack(out, dim=0)


class OnlineRLDataset(Dataset):
    """Prompt-only dataset for online RLHF (prompt_pg / prompt_awr / PPO):
    each item is a formatted prompt string ready for generation."""

    def __init__(
        self,
        dataset: Iterable[Dict],
        tokenizer,
        input_key: str = "input",
        apply_chat_template: bool = False,
        input_template: str = None,
    ) -> None:
        super().__init__()
        self.tokenizer = tokenizer
        template_fn = tokenizer.apply_chat_template if apply_chat_template else None
        self.prompts = [
            self._format(d, input_key, input_template, template_fn) for d in dataset
        ]

    @staticmethod
    def _format(data: Dict, input_key: str, input_template: str, template_fn) -> str:
        raw = data[input_key]
        if template_fn is not None:
            return template_fn(raw, tokenize=False, add_generation_prompt=True)
        return input_template.format(raw) if input_template else raw

    def __len__(self) -> int:
        return len(self.prompts)

    def __getitem__(self, idx: int) -> str:
        return self.prompts[idx]


class OfflineRLDataset(Dataset):
    """Prompt + response + scalar label dataset for offline RLHF (KTO/DPO).

    ``collate_fn`` tokenizes prompt+response jointly, enforces a trailing
    EOS, and left-pads the batch; ``prompt_ids_len`` lets the loss mask the
    prompt part.
    """

    def __init__(
        self,
        dataset: Iterable[Dict],
        tokenizer,
        max_length: int,
        input_key: str = "input",
        extra_input_keys: List[str] = None,
        output_key: str = "output",
        label_key: str = "label",
        apply_chat_template: bool = False,
        tokenizer_chat_template: str = None,
        input_template: str = None,
        num_processors: int = 8,
        parallel_load: bool = False,
    ) -> None:
        super().__init__()
        self.tokenizer = tokenizer
        self.max_length = max_length
        self.extra_input_keys = list(extra_input_keys or [])
        template_fn = tokenizer.apply_chat_template if apply_chat_template else None

        self.prompts: List[str] = []
        self.responses: List[str] = []
        self.labels: List[Any] = []
        self.prompt_ids_lens: List[int] = []
        for key in self.extra_input_keys:
            setattr(self, key, [])

        for data in dataset:
            item = self._build(data, input_key, output_key, label_key, input_template, template_fn)
            if item is None:
                continue  # prompt alone already fills the context window
            self.prompts.append(item['prompt'])
            self.responses.append(item['response'])
            self.labels.append(item['label'])
            self.prompt_ids_lens.append(item['prompt_ids_len'])
            for key in self.extra_input_keys:
                getattr(self, key).append(data[key])

    def _build(self, data, input_key, output_key, label_key, input_template, template_fn):
        if template_fn is not None:
            if output_key:
                prompt = template_fn(data[input_key], tokenize=False, add_generation_prompt=True)
                full = template_fn(data[input_key] + data[output_key], tokenize=False)
            else:  # chat list whose last turn is the response
                prompt = template_fn(data[input_key][:-1], tokenize=False, add_generation_prompt=True)
                full = template_fn(data[input_key], tokenize=False)
            response = full[len(prompt):]
        else:
            prompt = input_template.format(data[input_key]) if input_template else data[input_key]
            response = data[output_key]
        tok = self.tokenizer(
            prompt, max_length=self.max_length, padding=False, truncation=True,
            return_tensors="pt", add_special_tokens=False
        )
        n_prompt = int(tok["attention_mask"].sum())
        if n_prompt >= self.max_length - 2:
            return None
        return {
            'prompt': prompt,
            'response': response,
            'label': data[label_key],
            'prompt_ids_len': n_prompt,
        }

    def __len__(self) -> int:
        return len(self.prompts)

    def __getitem__(self, idx: int) -> Dict[str, Union[torch.Tensor, int]]:
        item = {
            "prompt": self.prompts[idx],
            "response": self.responses[idx],
            "label": self.labels[idx],
            "prompt_ids_len": self.prompt_ids_lens[idx],
        }
        for key in self.extra_input_keys:
            item[key] = getattr(self, key)[idx]
        return item

    def _encode_pair(self, prompt: str, response: str):
        text = (prompt + response).rstrip("\n")
        eos = getattr(self.tokenizer, 'eos_token', None)
        if eos and not text.endswith(eos):
            text = text + " " + eos
        enc = self.tokenizer(
            text, max_length=self.max_length, padding=False, truncation=True,
            return_tensors="pt", add_special_tokens=False
        )
        ids, mask = enc["input_ids"], enc["attention_mask"]
        # truncation may have cut the terminator: force it back
        ids[0, -1] = self.tokenizer.eos_token_id
        mask[0, -1] = 1
        return ids, mask

    def collate_fn(self, item_list: List[Dict[str, Union[torch.Tensor, int]]]) -> Dict[str, Any]:
        ids, masks, labels, plens = [], [], [], []
        extras: Dict[str, list] = {k: [] for k in self.extra_input_keys}
        for item in item_list:
            i, m = self._encode_pair(item["prompt"], item["response"])
            ids.append(i)
            masks.append(m)
            labels.append(item["label"])
            plens.append(item["prompt_ids_len"])
            for k in self.extra_input_keys:
                extras[k].append(item[k])
        batch = {
            "input_ids": zero_pad_sequences(ids, side="left",
                                            value=getattr(self.tokenizer, 'pad_token_id', 0) or 0).squeeze(1),
            "attention_mask": zero_pad_sequences(masks, side="left").squeeze(1),
            "label": torch.as_tensor(labels),
            "prompt_ids_len": torch.as_tensor(plens, dtype=torch.long),
        }
        for k, v in extras.items():
            batch[k] = v
        return batch
