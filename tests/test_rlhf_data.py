"""RLHF dataset tests with a local stub tokenizer (no hub access offline)."""
import torch


class StubTokenizer:
    """Whitespace tokenizer with a HF-compatible calling convention."""
    eos_token = "<eos>"
    eos_token_id = 1
    pad_token_id = 0

    def __call__(self, text, max_length=None, padding=False, truncation=False,
                 return_tensors=None, add_special_tokens=True):
        ids = [hash(w) % 1000 + 10 for w in text.split()]
        if truncation and max_length is not None:
            ids = ids[:max_length]
        t = torch.tensor([ids], dtype=torch.long)
        return {"input_ids": t, "attention_mask": torch.ones_like(t)}

    def apply_chat_template(self, messages, tokenize=False, add_generation_prompt=False):
        text = " ".join(m["content"] for m in messages)
        if add_generation_prompt:
            text += " <assistant>"
        return text


def test_zero_pad_sequences():
    from ding.utils.data import zero_pad_sequences
    seqs = [torch.ones(3, dtype=torch.long), torch.ones(5, dtype=torch.long)]
    left = zero_pad_sequences(seqs, side="left")
    right = zero_pad_sequences(seqs, side="right")
    assert left.shape == (2, 5) and right.shape == (2, 5)
    assert left[0, :2].sum() == 0 and right[0, 3:].sum() == 0


def test_online_rl_dataset():
    from ding.utils.data import OnlineRLDataset
    data = [{"input": f"question {i}"} for i in range(5)]
    ds = OnlineRLDataset(data, StubTokenizer(), input_template="Q: {} A:")
    assert len(ds) == 5
    assert ds[0] == "Q: question 0 A:"
    chat = [{"input": [{"role": "user", "content": "hi"}]}]
    ds2 = OnlineRLDataset(chat, StubTokenizer(), apply_chat_template=True)
    assert ds2[0].endswith("<assistant>")


def test_offline_rl_dataset_and_collate():
    from ding.utils.data import OfflineRLDataset
    data = [
        {"input": "what is two plus two", "output": " four", "label": 1},
        {"input": "capital of france", "output": " paris is the capital", "label": 0},
        {"input": " ".join(["verylong"] * 64), "output": " x", "label": 1},  # dropped: prompt fills window
    ]
    ds = OfflineRLDataset(data, StubTokenizer(), max_length=16)
    assert len(ds) == 2, "over-length prompt must be filtered"
    item = ds[0]
    assert set(item) >= {"prompt", "response", "label", "prompt_ids_len"}
    batch = ds.collate_fn([ds[0], ds[1]])
    assert batch["input_ids"].shape == batch["attention_mask"].shape
    assert batch["input_ids"].dim() == 2 and batch["input_ids"].shape[0] == 2
    assert batch["label"].tolist() == [1, 0]
    # every row ends with EOS (left padding keeps the tail aligned)
    assert (batch["input_ids"][:, -1] == StubTokenizer.eos_token_id).all()


def test_offline_rl_dataset_extra_keys():
    from ding.utils.data import OfflineRLDataset
    data = [{"input": "a b", "output": " c", "label": 1, "image": torch.zeros(2, 2)}]
    ds = OfflineRLDataset(data, StubTokenizer(), max_length=16, extra_input_keys=["image"])
    batch = ds.collate_fn([ds[0]])
    assert isinstance(batch["image"], list) and batch["image"][0].shape == (2, 2)
