"""Distributed dataloader helpers (reference: deepspeed/runtime/dataloader.py)."""

import torch
from torch.utils.data import DataLoader, DistributedSampler


class RepeatingLoader:
    """Wraps a loader to restart automatically at StopIteration
    (reference RepeatingLoader)."""

    def __init__(self, loader):
        self.loader = loader
        self.data_iter = iter(loader)

    def __iter__(self):
        return self

    def __len__(self):
        return len(self.loader)

    def __next__(self):
        try:
            return next(self.data_iter)
        except StopIteration:
            self.data_iter = iter(self.loader)
            return next(self.data_iter)


def build_dataloader(dataset, batch_size, world_size=1, rank=0, num_workers=0,
                     collate_fn=None, shuffle=False, drop_last=True):
    sampler = None
    if world_size > 1:
        sampler = DistributedSampler(dataset, num_replicas=world_size,
                                     rank=rank, shuffle=shuffle)
    return DataLoader(dataset, batch_size=batch_size, sampler=sampler,
                      shuffle=(shuffle and sampler is None),
                      num_workers=num_workers, collate_fn=collate_fn,
                      drop_last=drop_last, pin_memory=torch.cuda.is_available())


class SyntheticTextDataset(torch.utils.data.Dataset):
    """Random token ids + next-token labels of a fixed shape (there is no
    network for real datasets; BASELINE.json mandates synthetic data)."""

    def __init__(self, num_samples, seq_len, vocab_size, seed=1234):
        g = torch.Generator().manual_seed(seed)
        self.data = torch.randint(0, vocab_size, (num_samples, seq_len + 1),
                                  generator=g)

    def __len__(self):
        return self.data.shape[0]

    def __getitem__(self, idx):
        row = self.data[idx]
        return row[:-1], row[1:]
