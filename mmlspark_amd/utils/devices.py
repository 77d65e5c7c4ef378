"""Device selection helpers."""
from __future__ import annotations

import torch


def default_device(pref: str = "auto") -> torch.device:
    if pref in (None, "", "auto"):
        return torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    return torch.device(pref)
