"""MPT decoder-only causal LM, MI355X-native.

Behavioral parity with the reference's model family (llm-foundry
``mpt_causal_lm`` as configured by photon/conf/llm_config/mpt-*.yaml:
pre-LN decoder, ALiBi position bias, tied embedding/LM head, vocab 50368,
GELU MLP with expansion_ratio 4). Parameter names intentionally contain the
``transformer`` prefix and match the reference's module naming so that the
sorted-name wire format and the ``transformer`` filter key contract survive
(photon/utils.py:640-670).

The hot ops dispatch through photon_amd.ops: hand-written CDNA4 HIP kernels
on GPU (flash attention with fused ALiBi, fused LayerNorm, fused
cross-entropy), plain PyTorch on CPU (the reference's ``attn_impl: torch``
fallback, photon_llm_125M.sh:121).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.attention import alibi_slopes, flash_attention, flash_attention_qkv
from ..ops.cross_entropy import fused_cross_entropy
from ..ops.layernorm import FusedLayerNorm
from ..ops.linear import lt_available, lt_linear, lt_mlp


@dataclass
class MPTConfig:
    d_model: int = 768
    n_heads: int = 12
    n_layers: int = 12
    expansion_ratio: int = 4
    max_seq_len: int = 2048
    vocab_size: int = 50368
    attn_impl: str = "flash"  # "flash" (HIP kernel) | "torch" (SDPA fallback)
    alibi_bias_max: float = 8.0
    no_bias: bool = False
    init_std: float = 0.02
    loss_impl: str = "fused"  # "fused" (HIP CE kernel) | "torch"

    @property
    def d_head(self) -> int:
        return self.d_model // self.n_heads

    @classmethod
    def from_cfg(cls, model_cfg: dict) -> "MPTConfig":
        attn = model_cfg.get("attn_config", {}) or {}
        return cls(
            d_model=int(model_cfg["d_model"]),
            n_heads=int(model_cfg["n_heads"]),
            n_layers=int(model_cfg["n_layers"]),
            expansion_ratio=int(model_cfg.get("expansion_ratio", 4)),
            max_seq_len=int(model_cfg.get("max_seq_len", 2048)),
            vocab_size=int(model_cfg.get("vocab_size", 50368)),
            attn_impl=str(attn.get("attn_impl", "flash")),
            no_bias=bool(model_cfg.get("no_bias", False)),
        )


class MPTAttention(nn.Module):
    """Causal multi-head self-attention with ALiBi.

    Weights follow the reference naming: a single fused ``Wqkv`` projection
    and ``out_proj`` (llm-foundry MPT attention layout).
    """

    def __init__(self, cfg: MPTConfig):
        super().__init__()
        self.cfg = cfg
        bias = not cfg.no_bias
        # instance attrs (not cfg reads) so tensor parallelism can shard the
        # head dimension per rank (photon_amd.parallel.tp).
        self.n_heads = cfg.n_heads
        self.d_head = cfg.d_head
        self.Wqkv = nn.Linear(cfg.d_model, 3 * cfg.d_model, bias=bias)
        self.out_proj = nn.Linear(cfg.d_model, cfg.d_model, bias=bias)
        slopes = alibi_slopes(cfg.n_heads, cfg.alibi_bias_max)
        self.register_buffer("slopes", slopes, persistent=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, S, D = x.shape
        H, dh = self.n_heads, self.d_head
        # Fused-linear path (C++ autograd + HIP bias-grad backward). Only
        # for PLAIN nn.Linear projections: tensor-parallel swaps replace
        # them with Column/RowParallelLinear whose forward carries the TP
        # collectives — bypassing those with raw .weight would silently
        # drop the all-reduces.
        fused = (
            lt_available(x)
            and type(self.Wqkv) is nn.Linear
            and type(self.out_proj) is nn.Linear
        )
        qkv = (
            lt_linear(x, self.Wqkv.weight, self.Wqkv.bias)
            if fused
            else self.Wqkv(x)
        )
        # Packed path: the HIP kernels read [B,S,3,H,dh] strided directly
        # (no chunk/transpose/contiguous copies); falls back to reshape +
        # SDPA on CPU / attn_impl=torch.
        out = flash_attention_qkv(
            qkv, H, self.slopes, causal=True, impl=self.cfg.attn_impl
        )  # [B, S, H*dh]
        if fused:
            return lt_linear(out, self.out_proj.weight, self.out_proj.bias)
        return self.out_proj(out)


class MPTMLP(nn.Module):
    def __init__(self, cfg: MPTConfig):
        super().__init__()
        bias = not cfg.no_bias
        hidden = cfg.expansion_ratio * cfg.d_model
        self.up_proj = nn.Linear(cfg.d_model, hidden, bias=bias)
        # tanh flavor matches the hipBLASLt GELU epilogue (diff vs exact
        # GELU is below bf16 resolution; keeps CPU/GPU paths comparable)
        self.act = nn.GELU(approximate="tanh")
        self.down_proj = nn.Linear(hidden, cfg.d_model, bias=bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # plain-Linear guard: see MPTAttention.forward (TP swap safety)
        if (lt_available(x) and type(self.up_proj) is nn.Linear
                and type(self.down_proj) is nn.Linear):
            return lt_mlp(x, self.up_proj.weight, self.up_proj.bias,
                          self.down_proj.weight, self.down_proj.bias)
        return self.down_proj(self.act(self.up_proj(x)))


class MPTBlock(nn.Module):
    def __init__(self, cfg: MPTConfig):
        super().__init__()
        bias = not cfg.no_bias
        self.norm_1 = FusedLayerNorm(cfg.d_model, bias=bias)
        self.attn = MPTAttention(cfg)
        self.norm_2 = FusedLayerNorm(cfg.d_model, bias=bias)
        self.ffn = MPTMLP(cfg)

    def forward(self, x: torch.Tensor, pending: torch.Tensor | None = None):
        """Residual-add/LN fusion flow: ``pending`` is the PREVIOUS
        block's un-added FFN output (or None for block 0); the add rides
        the norm_1 kernel. Returns (residual_stream, ffn_out) — the caller
        (MPTModel) threads ffn_out into the next block / the final norm.
        Eager semantics are identical: x+pending -> norm_1 -> attn -> add
        -> norm_2 -> ffn."""
        if pending is not None:
            x, n1 = self.norm_1.forward_add(x, pending)
        else:
            n1 = self.norm_1(x)
        a = self.attn(n1)
        x, n2 = self.norm_2.forward_add(x, a)
        return x, self.ffn(n2)


class MPTModel(nn.Module):
    """The ``transformer`` trunk: wte + blocks + norm_f (no positional
    embedding — ALiBi handles position)."""

    def __init__(self, cfg: MPTConfig):
        super().__init__()
        self.wte = nn.Embedding(cfg.vocab_size, cfg.d_model)
        self.blocks = nn.ModuleList(MPTBlock(cfg) for _ in range(cfg.n_layers))
        self.norm_f = FusedLayerNorm(cfg.d_model, bias=not cfg.no_bias)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        x = self.wte(input_ids)
        # Keep the residual stream in the autocast compute dtype (bf16):
        # nn.Embedding outputs fp32 under autocast, and without this cast
        # every LayerNorm / residual add / attention input runs fp32 —
        # doubling HBM traffic on a bandwidth-bound chip (measured 16.6% of
        # step time in fp32 LN alone, profiles/r01). Statistics inside the
        # LN/CE kernels still accumulate in fp32.
        if torch.is_autocast_enabled(x.device.type):
            x = x.to(torch.get_autocast_dtype(x.device.type))
        pending = None
        for block in self.blocks:
            x, pending = block(x, pending)
        # final residual add fused into norm_f
        _, y = self.norm_f.forward_add(x, pending)
        return y


class MPTCausalLM(nn.Module):
    """MPT causal LM with tied wte/LM-head (llm-foundry behavior)."""

    def __init__(self, cfg: MPTConfig):
        super().__init__()
        self.cfg = cfg
        self.transformer = MPTModel(cfg)
        self.apply(self._init_weights)

    def _init_weights(self, module: nn.Module) -> None:
        std = self.cfg.init_std
        if isinstance(module, nn.Linear):
            nn.init.normal_(module.weight, mean=0.0, std=std)
            if module.bias is not None:
                nn.init.zeros_(module.bias)
        elif isinstance(module, nn.Embedding):
            nn.init.normal_(module.weight, mean=0.0, std=std)
        elif isinstance(module, FusedLayerNorm):
            nn.init.ones_(module.weight)
            if module.bias is not None:
                nn.init.zeros_(module.bias)

    def logits(self, input_ids: torch.Tensor) -> torch.Tensor:
        h = self.transformer(input_ids)
        return F.linear(h, self.transformer.wte.weight)

    def forward(
        self, input_ids: torch.Tensor, labels: torch.Tensor | None = None
    ) -> dict[str, torch.Tensor]:
        h = self.transformer(input_ids)
        if labels is None:
            return {"logits": F.linear(h, self.transformer.wte.weight)}
        # Shift so token t predicts token t+1 (causal LM convention).
        h = h[:, :-1, :]
        tgt = labels[:, 1:]
        loss = fused_cross_entropy(
            h.reshape(-1, h.shape[-1]),
            self.transformer.wte.weight,
            tgt.reshape(-1),
            impl=self.cfg.loss_impl,
        )
        return {"loss": loss}

    @torch.no_grad()
    def num_params(self, trainable_only: bool = True) -> int:
        return sum(
            p.numel()
            for p in self.parameters()
            if (p.requires_grad or not trainable_only)
        )


@torch.no_grad()
def resize_vocab(model: MPTCausalLM, new_vocab: int) -> MPTCausalLM:
    """Resize the tied wte/LM-head vocabulary (reference fl.resize_vocab,
    base_schema.py FL): existing rows kept, new rows init-normal. Changes
    the wire format size — all participants must use the same value."""
    old = model.transformer.wte
    if new_vocab == old.num_embeddings:
        return model
    new = nn.Embedding(new_vocab, old.embedding_dim,
                       device=old.weight.device, dtype=old.weight.dtype)
    nn.init.normal_(new.weight, mean=0.0, std=model.cfg.init_std)
    keep = min(new_vocab, old.num_embeddings)
    new.weight.data[:keep] = old.weight.data[:keep]
    model.transformer.wte = new
    model.cfg.vocab_size = new_vocab
    return model


def build_model(llm_config: dict) -> MPTCausalLM:
    """Build from the llm_config subtree (photon_amd.conf)."""
    cfg = MPTConfig.from_cfg(llm_config["model"])
    return MPTCausalLM(cfg)
