"""Model configuration.

Mirrors the keyword surface of the reference ProGen factory
(reference: progen_transformer/progen.py:188-203,235) plus the TOML config
loading convention (reference: train.py:95-98, configs/model/default.toml).
"""

from __future__ import annotations

import dataclasses
from pathlib import Path
from typing import Any, Dict, Optional

try:
    import tomllib  # py311+
except ModuleNotFoundError:  # py310: tomli is the same parser
    import tomli as tomllib


@dataclasses.dataclass
class ProGenConfig:
    """Hyperparameters of a ProGen model.

    Field set and defaults match the reference ProGenBase constructor
    (reference: progen_transformer/progen.py:188-203). ``attn_dim`` and
    ``clamp_gate`` are accepted but unused there too (dead kwargs,
    reference: progen.py:201-202) — kept for checkpoint/config parity.
    """

    num_tokens: int = 256
    dim: int = 512
    seq_len: int = 1024
    depth: int = 6
    window_size: int = 256
    global_mlp_depth: int = 2
    heads: int = 8
    dim_head: int = 64
    ff_mult: int = 4
    ff_glu: bool = True
    attn_dim: Optional[int] = None   # dead kwarg (parity)
    clamp_gate: bool = True          # dead kwarg (parity)
    shift_tokens: bool = True
    # --- MI355X-native extensions (not in reference) ---
    # compute dtype for GPU training ("bf16" | "fp32")
    compute_dtype: str = "bf16"

    def __post_init__(self) -> None:
        if self.seq_len % self.window_size != 0:
            raise ValueError(
                f"seq_len ({self.seq_len}) must be divisible by window_size "
                f"({self.window_size})"  # reference: progen.py:80
            )

    @property
    def inner_dim(self) -> int:
        return self.heads * self.dim_head

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ProGenConfig":
        known = {f.name for f in dataclasses.fields(cls)}
        # tolerate reference checkpoints / configs carrying extra keys
        return cls(**{k: v for k, v in d.items() if k in known})

    @classmethod
    def from_toml(cls, path: str | Path) -> "ProGenConfig":
        return cls.from_dict(tomllib.loads(Path(path).read_text()))

    def to_dict(self) -> Dict[str, Any]:
        return dataclasses.asdict(self)
