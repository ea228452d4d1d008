"""Sort-task dataset (parity: reference examples/sorter/dataset.py):
input = 6 random digits, target = the sorted sequence; sequences are
(input | sorted) with loss masked to the output half (ignore_index=-1)."""
import torch


class SortDataset:
    def __init__(self, split="train", length=6, num_digits=3, size=10000,
                 seed=3407):
        self.length = length
        self.num_digits = num_digits
        g = torch.Generator().manual_seed(seed + (0 if split == "train" else 1))
        self.data = torch.randint(num_digits, (size, length), generator=g)

    def __len__(self):
        return len(self.data)

    def __getitem__(self, idx):
        inp = self.data[idx]
        sol = torch.sort(inp)[0]
        cat = torch.cat((inp, sol), dim=0)
        x = cat[:-1].clone()
        y = cat[1:].clone()
        y[:self.length - 1] = -1  # only predict the sorted half
        return x, y

    @property
    def vocab_size(self):
        return self.num_digits

    @property
    def block_size(self):
        return self.length * 2 - 1
