# Model/training configuration for the hot path.
#
# Mirrors the reference's configs: examples/gpt2.yaml (GPT-2 small,
# microbatch 8, global microbatch 128) and examples/gpt3.yaml (GPT-2 XL dims:
# n_embd=1600, 48 layers, 25 heads, microbatch 2) —
# /root/reference/examples/*.yaml, elastic/training_util.py:7-39.
from __future__ import annotations

from dataclasses import dataclass


@dataclass
class ModelConfig:
    n_embd: int = 768
    n_head: int = 12
    n_layer: int = 12            # transformer blocks
    n_positions: int = 1024
    vocab_size: int = 50257

    @property
    def n_layers_total(self) -> int:
        # fx-shard grain: embedding + blocks + (ln_f + lm_head + loss)
        # (/root/reference/oobleck/module/sharding.py:12-47)
        return self.n_layer + 2

    def layer_kind(self, layer_id: int) -> int:
        from .params import KIND_BLOCK, KIND_EMBED, KIND_FINAL
        if layer_id == 0:
            return KIND_EMBED
        if layer_id == self.n_layers_total - 1:
            return KIND_FINAL
        return KIND_BLOCK


GPT2_SMALL = ModelConfig()
# examples/gpt3.yaml model_args (the yaml's actual dims; BASELINE.json calls
# this "GPT-3 2.7B", SURVEY.md §5 notes it is 1600x48 ≈ GPT-2 XL 1.56B)
GPT2_XL = ModelConfig(n_embd=1600, n_head=25, n_layer=48, n_positions=1024)


@dataclass
class TrainingConfig:
    microbatch_size: int = 8          # gpt2.yaml:5
    global_microbatch_size: int = 128  # gpt2.yaml:6
    seq_len: int = 1024
    lr: float = 5e-5                  # HF TrainingArguments default
    adam_beta1: float = 0.9
    adam_beta2: float = 0.999
    adam_eps: float = 1e-8
    weight_decay: float = 0.0
    warmup_steps: int = 0

    @property
    def num_microbatches(self) -> int:
        assert self.global_microbatch_size % self.microbatch_size == 0
        return self.global_microbatch_size // self.microbatch_size


def synthetic_batch(cfg: ModelConfig, batch: int, seq: int, seed: int):
    """Synthetic token batch (no network: dataset.py:150-208's tokenized
    wikitext is replaced by seeded random ids; labels = input_ids copies the
    reference's dataset.py:201)."""
    import torch
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, cfg.vocab_size, (batch, seq), generator=g)
    return ids, ids.clone()
