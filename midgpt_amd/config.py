"""Experiment / model configuration dataclasses.

Field names and semantics mirror the reference's ``ExperimentConfig``
(reference src/train.py:26-44) and ``GPTConfig`` (src/model.py:108-115)
so that configs and the frozen ``config.json`` in a rundir are
interchangeable at the contract level.
"""
from __future__ import annotations

import dataclasses
import json
from dataclasses import dataclass
from typing import Optional


@dataclass
class GPTConfig:
    block_size: int          # Max sequence length T
    vocab_size: int          # Number of tokens V
    n_layer: int             # Number of transformer blocks L
    n_head: int              # Number of attention heads H
    n_embd: int              # Hidden dimension D
    dropout: float

    @property
    def head_dim(self) -> int:
        return self.n_embd // self.n_head


@dataclass
class ExperimentConfig:
    rundir: str              # Directory containing ckpts and logs.
    data_dir: str            # Dataset directory
    learning_rate: float
    batch_size: int          # GLOBAL batch size across all devices
    warmup_steps: int
    min_lr: float            # Final LR after decay
    lr_decay_steps: int
    max_steps: int           # Number of optimizer steps
    beta2: float
    weight_decay: float
    eval_interval: int
    param_dtype: str         # 'float32' (master weights)
    compute_dtype: str       # 'bfloat16' or 'float32'
    g_accum_iters: int       # Gradient accumulation microbatches per step
    shard_model: bool        # ZeRO-3-style param sharding (vs ZeRO-1 resident)
    model_config: GPTConfig
    debug: bool = False
    # --- extensions over the reference contract (defaulted, so reference
    # configs load unchanged) ---
    seed: Optional[int] = None        # opt-in seeded data sampling (ref is unseeded)
    beta1: float = 0.9                # optax scale_by_adam default b1
    grad_clip: float = 1.0            # optax clip_by_global_norm(1.0)
    adam_eps: float = 1e-8            # optax scale_by_adam default eps
    remat: bool = True                # per-block activation recompute (jax.checkpoint parity)
    synthetic_data: bool = False      # random tokens instead of data_dir bins (benchmarks)

    def to_json(self) -> str:
        d = dataclasses.asdict(self)
        return json.dumps(d, indent=2)

    @staticmethod
    def from_json(s: str) -> "ExperimentConfig":
        d = json.loads(s)
        mc = d.pop("model_config")
        known_m = {f.name for f in dataclasses.fields(GPTConfig)}
        known_e = {f.name for f in dataclasses.fields(ExperimentConfig)} - {"model_config"}
        model = GPTConfig(**{k: v for k, v in mc.items() if k in known_m})
        return ExperimentConfig(model_config=model,
                                **{k: v for k, v in d.items() if k in known_e})


def load_config(name: str) -> ExperimentConfig:
    """Dynamic config import by module name, mirroring launch.py:25-27."""
    import importlib
    try:
        mod = importlib.import_module(f"midgpt_amd.configs.{name}")
    except ModuleNotFoundError as e:
        import pkgutil
        import midgpt_amd.configs as cfgs
        avail = sorted(m.name for m in pkgutil.iter_modules(cfgs.__path__)
                       if not m.name.startswith("_"))
        raise SystemExit(f"unknown config '{name}' ({e}); available: "
                         f"{', '.join(avail)}")
    return mod.config
