"""Dataset utilities (parity: reference utils/data.py).

``RolloverTensorDataset`` repeats a short input sequence up to a requested
length (reference data.py:7-20). Dataset loaders: synthetic (offline default
for this environment - no network), plus CoLA/GLUE and ImageNet loaders that
require local data and optional deps (``datasets`` / ``torchvision``), gated
at call time like the reference (data.py:54-89).
"""
from typing import Optional, Sequence

import torch
from torch.utils.data import Dataset


class RolloverTensorDataset(Dataset):
    """Wraps tensors, repeating entries (with rollover) up to ``length``."""

    def __init__(self, length: int, *tensors: torch.Tensor):
        assert len(tensors) > 0
        assert all(t.shape[0] == tensors[0].shape[0] for t in tensors)
        self.tensors = tensors
        self.length = length

    def __len__(self):
        return self.length

    def __getitem__(self, idx):
        if idx >= self.length:
            raise IndexError
        base = idx % self.tensors[0].shape[0]
        return tuple(t[base] for t in self.tensors)


class ShuffledDataset(Dataset):
    """Index-permuted view of a dataset (``--dataset-shuffle``; the reference
    shuffles at load time, runtime.py:358-401)."""

    def __init__(self, base: Dataset, seed: int = 0):
        self.base = base
        self.perm = torch.randperm(
            len(base), generator=torch.Generator().manual_seed(seed)).tolist()

    def __len__(self):
        return len(self.base)

    def __getitem__(self, idx):
        return self.base[self.perm[idx]]


def load_dataset_synthetic(model_config, batch_size: int,
                           seq_len: Optional[int] = None,
                           seed: int = 0) -> Dataset:
    """Random inputs shaped for the model (ImageNet-shaped pixels or token
    ids), with random labels - the offline benchmarking default."""
    gen = torch.Generator().manual_seed(seed)
    if model_config.model_type in ("vit", "deit"):
        data = torch.randn(batch_size, model_config.num_channels,
                           model_config.image_size, model_config.image_size,
                           generator=gen)
        labels = torch.randint(0, max(model_config.num_labels, 1),
                               (batch_size,), generator=gen)
    else:
        s = seq_len or model_config.max_position_embeddings
        # ids start past the special-token range so no synthetic position
        # collides with pad_token_id (full-length sequences, no padding)
        lo = min(1000, model_config.vocab_size - 1)
        data = torch.randint(lo, model_config.vocab_size, (batch_size, s),
                             generator=gen)
        labels = torch.randint(0, max(model_config.num_labels, 2),
                               (batch_size,), generator=gen)
    return RolloverTensorDataset(batch_size, data, labels)


def load_dataset_glue(model_name: str, dataset_name: str, split: str,
                      batch_size: int, tokenizer_name: str = "bert-base-uncased",
                      indices: Optional[Sequence[int]] = None) -> Dataset:
    """GLUE dataset via HF ``datasets`` (requires local cache; reference
    data.py:54-78)."""
    from datasets import load_dataset as hf_load  # noqa: PLC0415
    from transformers import BertTokenizer  # noqa: PLC0415
    dset = hf_load("glue", dataset_name.lower(), split=split)
    tokenizer = BertTokenizer.from_pretrained(tokenizer_name)
    sentences = dset["sentence"][:batch_size]
    labels = torch.tensor(dset["label"][:batch_size])
    enc = tokenizer(sentences, padding=True, truncation=True,
                    return_tensors="pt")
    data = enc["input_ids"]
    if indices is not None:
        data = data[list(indices)]
        labels = labels[list(indices)]
    return RolloverTensorDataset(batch_size, data, labels)


def load_dataset_imagenet(root: str, split: str, batch_size: int,
                          indices: Optional[Sequence[int]] = None) -> Dataset:
    """ImageNet via torchvision (requires local archives; reference
    data.py:80-89)."""
    from torchvision import transforms  # noqa: PLC0415
    from torchvision.datasets import ImageNet  # noqa: PLC0415
    tform = transforms.Compose([
        transforms.Resize(256), transforms.CenterCrop(224),
        transforms.ToTensor(),
        transforms.Normalize(mean=[0.485, 0.456, 0.406],
                             std=[0.229, 0.224, 0.225]),
    ])
    dset = ImageNet(root, split=split, transform=tform)
    data, labels = [], []
    idxs = indices if indices is not None else range(batch_size)
    for i in idxs:
        img, lbl = dset[i]
        data.append(img)
        labels.append(lbl)
        if len(data) >= batch_size:
            break
    return RolloverTensorDataset(batch_size, torch.stack(data),
                                 torch.tensor(labels))


def load_dataset_npz(path: str, batch_size: int) -> Dataset:
    """Tokenized-input dataset from an ``.npz`` file (offline analogue of
    the reference's tools/bert_save_input.py flow, which stores raw CoLA
    sentences and tokenizes at load - tokenizer vocabularies need network
    access, so this environment stores token ids directly).

    Keys: ``ids`` int64 [N, S] (padded with the model's pad id) and
    optional ``label`` [N].
    """
    import numpy as np
    data = np.load(path, allow_pickle=True)
    key = "ids" if "ids" in data else "input"
    ids = torch.as_tensor(np.asarray(data[key], dtype="int64"))
    if "label" in data:
        labels = torch.as_tensor(np.asarray(data["label"], dtype="int64"))
    else:
        labels = torch.zeros(ids.shape[0], dtype=torch.int64)
    return RolloverTensorDataset(batch_size, ids, labels)
