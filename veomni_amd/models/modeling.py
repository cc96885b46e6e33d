"""Decoder-only modeling for the BASELINE model families, written against the
OpSlot operator API (no HF dependency).

Covers: Llama-3 / Qwen2(-0.5B) dense (optional attention bias, no q/k norm)
and Qwen3-MoE (per-head q/k RMSNorm, top-k router + experts). Math order
mirrors the reference's generated modeling so CPU eager runs are
bit-comparable (anchors: /root/reference/veomni/models/transformers/qwen3_moe/
generated/patched_modeling_qwen3_moe_gpu.py — RMSNorm :380-400, RoPE :94-111,
attention :150-220, MoE block :338-370, experts :254-294, router :303-330,
decoder layer :405-450). Parameter names are HF-compatible so state dicts
interoperate.

OpSlots (bound by `bind_ops`, ref auto.py:63-107 `_bind_veomni_ops`):
  rms_norm/standard, rotary_pos_emb/full, swiglu_mlp/standard,
  moe_experts/standard, cross_entropy_loss/causal, load_balancing_loss/standard
plus the attention interface (veomni_flash_attention_2_with_sp semantics) via
`veomni_amd.ops.kernels.attention`.
"""

from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..distributed.parallel_state import get_parallel_state
from ..ops.dispatch import OpSlot

veomni_rms_norm = OpSlot("rms_norm", "standard")
veomni_apply_rotary_pos_emb = OpSlot("rotary_pos_emb", "full")
veomni_swiglu_mlp = OpSlot("swiglu_mlp", "standard")
veomni_moe_experts_forward = OpSlot("moe_experts", "standard")
veomni_causal_lm_loss = OpSlot("cross_entropy_loss", "causal")
veomni_load_balancing_loss = OpSlot("load_balancing_loss", "standard")
veomni_attention = OpSlot("attention", "sdpa_with_sp")


@dataclass
class ModelConfig:
    vocab_size: int = 151936
    hidden_size: int = 2048
    intermediate_size: int = 6144
    num_hidden_layers: int = 4
    num_attention_heads: int = 32
    num_key_value_heads: int = 4
    head_dim: Optional[int] = None
    rms_norm_eps: float = 1e-6
    rope_theta: float = 1000000.0
    max_position_embeddings: int = 32768
    attention_bias: bool = False
    qk_norm: bool = False              # True for Qwen3(-MoE)
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02
    # MoE
    num_experts: int = 0               # 0 => dense MLP
    num_experts_per_tok: int = 8
    moe_intermediate_size: int = 768
    norm_topk_prob: bool = True
    router_aux_loss_coef: float = 0.0  # >0 enables the aux loss path
    # multimodal (Qwen2.5-VL) 3D-RoPE channel sections (t, h, w); None = 1D
    mrope_section: Optional[tuple] = None
    name: str = "model"

    def __post_init__(self):
        if self.head_dim is None:
            self.head_dim = self.hidden_size // self.num_attention_heads

    @property
    def is_moe(self) -> bool:
        return self.num_experts > 0


class RMSNorm(nn.Module):
    def __init__(self, hidden_size: int, eps: float = 1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size))
        self.variance_epsilon = eps

    def forward(self, hidden_states: torch.Tensor) -> torch.Tensor:
        if veomni_rms_norm.use_non_eager_impl:
            return veomni_rms_norm(hidden_states, self.weight, self.variance_epsilon)
        input_dtype = hidden_states.dtype
        x = hidden_states.to(torch.float32)
        variance = x.pow(2).mean(-1, keepdim=True)
        x = x * torch.rsqrt(variance + self.variance_epsilon)
        return self.weight * x.to(input_dtype)


def rotate_half(x):
    x1 = x[..., : x.shape[-1] // 2]
    x2 = x[..., x.shape[-1] // 2 :]
    return torch.cat((-x2, x1), dim=-1)


def apply_rotary_pos_emb(q, k, cos, sin, position_ids=None, unsqueeze_dim=1):
    if veomni_apply_rotary_pos_emb.use_non_eager_impl:
        return veomni_apply_rotary_pos_emb(q, k, cos, sin, position_ids=position_ids,
                                           unsqueeze_dim=unsqueeze_dim)
    cos = cos.unsqueeze(unsqueeze_dim)
    sin = sin.unsqueeze(unsqueeze_dim)
    return (q * cos) + (rotate_half(q) * sin), (k * cos) + (rotate_half(k) * sin)


class RotaryEmbedding(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        inv_freq = 1.0 / (
            config.rope_theta
            ** (torch.arange(0, config.head_dim, 2, dtype=torch.float32) / config.head_dim)
        )
        self.register_buffer("inv_freq", inv_freq, persistent=False)
        self.mrope_section = tuple(config.mrope_section) if config.mrope_section else None

    @torch.no_grad()
    def forward(self, x: torch.Tensor, position_ids: torch.Tensor):
        # position_ids [B, S] (1D rope) or [3, B, S] (Qwen2.5-VL mrope);
        # fp32 angles, output in x.dtype (HF default rope). For mrope the
        # per-axis cos/sin are SECTION-MERGED here once
        # (ref apply_multimodal_rotary_pos_emb, patched_modeling_qwen2_5_vl
        # _gpu.py:1027-1068: channel sections t/h/w, duplicated for the
        # rotate-half second half) — downstream attention then applies
        # STANDARD rope, so the HIP rope kernel serves the VLM unchanged.
        if position_ids.dim() == 3:
            inv = self.inv_freq[None, None, :, None].float().expand(
                3, position_ids.shape[1], -1, 1)
            pos = position_ids[:, :, None, :].float()
            freqs = (inv @ pos).transpose(2, 3)       # [3, B, S, D/2]
            emb = torch.cat((freqs, freqs), dim=-1)   # [3, B, S, D]
            cos, sin = emb.cos(), emb.sin()
            sec = list(self.mrope_section) * 2
            cos = torch.cat([m[i % 3] for i, m in enumerate(cos.split(sec, dim=-1))], dim=-1)
            sin = torch.cat([m[i % 3] for i, m in enumerate(sin.split(sec, dim=-1))], dim=-1)
            return cos.to(x.dtype), sin.to(x.dtype)
        inv = self.inv_freq[None, :, None].float().expand(position_ids.shape[0], -1, 1)
        pos = position_ids[:, None, :].float()
        freqs = (inv @ pos).transpose(1, 2)
        emb = torch.cat((freqs, freqs), dim=-1)
        return emb.cos().to(x.dtype), emb.sin().to(x.dtype)


def repeat_kv(hidden_states: torch.Tensor, n_rep: int) -> torch.Tensor:
    b, h, s, d = hidden_states.shape
    if n_rep == 1:
        return hidden_states
    return (
        hidden_states[:, :, None, :, :].expand(b, h, n_rep, s, d).reshape(b, h * n_rep, s, d)
    )


def sdpa_attention(module, query, key, value, attention_mask, dropout=0.0,
                   scaling=None, doc_start=None, **kwargs):
    """Eager/SDPA attention on [B,h,S,D] (HF sdpa semantics). GQA repeat is
    derived from the shapes (under Ulysses the per-rank head counts differ
    from the module's config attrs). doc_start (int32 [S]) applies the
    packed-varlen block-diagonal causal mask (the reference flash-attn
    cu_seqlens semantics, attention/flash.py:61-91) as an explicit mask —
    this is the eager oracle for the HIP varlen kernels."""
    n_rep = query.shape[1] // key.shape[1]
    key = repeat_kv(key, n_rep)
    value = repeat_kv(value, n_rep)
    if doc_start is not None:
        S = query.shape[2]
        assert doc_start.numel() == S, (doc_start.numel(), S)
        ar = torch.arange(S, device=query.device)
        ds = doc_start.to(device=query.device, dtype=torch.long)
        allowed = (ar[None, :] <= ar[:, None]) & (ar[None, :] >= ds[:, None])
        out = F.scaled_dot_product_attention(
            query, key, value, attn_mask=allowed, dropout_p=dropout,
            scale=scaling, is_causal=False
        )
    else:
        out = F.scaled_dot_product_attention(
            query, key, value, attn_mask=None, dropout_p=dropout,
            scale=scaling, is_causal=True
        )
    return out.transpose(1, 2).contiguous(), None


class Attention(nn.Module):
    def __init__(self, config: ModelConfig, layer_idx: int):
        super().__init__()
        self.config = config
        self.layer_idx = layer_idx
        self.head_dim = config.head_dim
        self.num_heads = config.num_attention_heads
        self.num_key_value_heads = config.num_key_value_heads
        self.num_key_value_groups = config.num_attention_heads // config.num_key_value_heads
        self.scaling = self.head_dim**-0.5
        self.is_causal = True
        H = config.hidden_size
        bias = config.attention_bias
        self.q_proj = nn.Linear(H, self.num_heads * self.head_dim, bias=bias)
        self.k_proj = nn.Linear(H, self.num_key_value_heads * self.head_dim, bias=bias)
        self.v_proj = nn.Linear(H, self.num_key_value_heads * self.head_dim, bias=bias)
        self.o_proj = nn.Linear(self.num_heads * self.head_dim, H, bias=False)
        if config.qk_norm:
            self.q_norm = RMSNorm(self.head_dim, eps=config.rms_norm_eps)
            self.k_norm = RMSNorm(self.head_dim, eps=config.rms_norm_eps)
        else:
            self.q_norm = self.k_norm = None

    def forward(self, hidden_states, position_embeddings, **kwargs):
        input_shape = hidden_states.shape[:-1]
        ps = get_parallel_state()
        if ps.ulysses_enabled and ps.async_ulysses:
            # deepened async Ulysses (ref async_ulysses.py:48-419): the qkv
            # GEMM is split back into per-tensor launches on the comm path
            # so each all-to-all overlaps the next projection
            out = self._async_ulysses_attention(hidden_states,
                                                position_embeddings, **kwargs)
            out = out.reshape(*input_shape, -1).contiguous()
            return self.o_proj(out)
        # one fused qkv GEMM (weights concatenated at call time, so the
        # per-projection parameters/state-dict stay reference-shaped); the
        # per-element math is identical to three separate linears.
        wqkv = torch.cat((self.q_proj.weight, self.k_proj.weight, self.v_proj.weight))
        bqkv = None
        if self.q_proj.bias is not None:
            bqkv = torch.cat((self.q_proj.bias, self.k_proj.bias, self.v_proj.bias))
        qkv = nn.functional.linear(hidden_states, wqkv, bqkv)
        hv = qkv.view(*input_shape, -1, self.head_dim)
        nh, nkv = self.num_heads, self.num_key_value_heads
        q = hv[..., :nh, :]
        k = hv[..., nh:nh + nkv, :]
        v = hv[..., nh + nkv:, :]
        if self.q_norm is not None:
            q = self.q_norm(q)
            k = self.k_norm(k)
        q, k, v = q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2)
        cos, sin = position_embeddings
        q, k = apply_rotary_pos_emb(q, k, cos, sin)
        if veomni_attention.use_non_eager_impl:
            # the bound attention kernel is SP-aware (does the Ulysses
            # exchange itself, ref attention/flash.py:236-299); varlen
            # doc bounds ride the kwargs (ref flash.py:61-91)
            out, _ = veomni_attention(self, q, k, v, None, dropout=0.0,
                                      scaling=self.scaling, **kwargs)
        elif ps.ulysses_enabled:
            # eager under SP still needs the sync Ulysses exchange
            out = self._sync_ulysses_attention(q, k, v, **kwargs)
        else:
            out, _ = sdpa_attention(self, q, k, v, None, dropout=0.0,
                                    scaling=self.scaling, **kwargs)
        out = out.reshape(*input_shape, -1).contiguous()
        return self.o_proj(out)

    def _sync_ulysses_attention(self, q, k, v, **kwargs):
        from ..distributed.sequence_parallel import (
            gather_heads_scatter_seq,
            gather_seq_scatter_heads,
        )

        ps = get_parallel_state()
        group = ps.ulysses_group
        sp = ps.ulysses_size
        assert q.shape[0] == 1, "ulysses expects packed batch (B == 1)"
        kv = k.shape[1]
        if sp > kv:
            rep = sp // kv
            k = torch.repeat_interleave(k, dim=1, repeats=rep)
            v = torch.repeat_interleave(v, dim=1, repeats=rep)
        qs = gather_seq_scatter_heads(q.squeeze(0).transpose(0, 1).contiguous(),
                                      seq_dim=0, head_dim=1, group=group)
        ks = gather_seq_scatter_heads(k.squeeze(0).transpose(0, 1).contiguous(),
                                      seq_dim=0, head_dim=1, group=group)
        vs = gather_seq_scatter_heads(v.squeeze(0).transpose(0, 1).contiguous(),
                                      seq_dim=0, head_dim=1, group=group)
        out, _ = sdpa_attention(self, qs.transpose(0, 1)[None], ks.transpose(0, 1)[None],
                                vs.transpose(0, 1)[None], None, dropout=0.0,
                                scaling=self.scaling, **kwargs)  # [1, S, h/sp, D]
        out = gather_heads_scatter_seq(out.squeeze(0), head_dim=1, seq_dim=0, group=group)
        return out[None]

    def _async_ulysses_attention(self, hidden_states, position_embeddings,
                                 **kwargs):
        """Async Ulysses comm path (ref async_ulysses.py:48-419): per-tensor
        projection -> qk-norm -> local-slice RoPE -> ASYNC seq->head a2a, in
        q/k/v order, so q's exchange flies during the k projection GEMM and
        k's during v's. The three waits then land just before core
        attention. Backward mirrors the overlap: _A2AWait.backward starts the
        reverse exchanges (all three back-to-back in autograd order) and
        _A2AStartSeqHeads.backward finishes each, so the v/k projection
        weight-grad GEMMs run under q/k's reverse a2a. RoPE commutes with the
        seq gather (elementwise per position), so it runs on the local slice;
        KV heads repeat before the a2a when sp > kv heads."""
        from ..distributed.sequence_parallel import (
            gather_heads_scatter_seq,
            gather_seq_scatter_heads_async,
            wait_gathered,
        )

        ps = get_parallel_state()
        sp = ps.ulysses_size
        group = ps.ulysses_group
        assert hidden_states.shape[0] == 1, "async ulysses expects packed batch (B == 1)"
        S_loc = hidden_states.shape[1]
        nh, nkv = self.num_heads, self.num_key_value_heads
        rep = sp // nkv if sp > nkv else 1
        cos, sin = position_embeddings
        cos_u, sin_u = cos.unsqueeze(1), sin.unsqueeze(1)

        q = self.q_proj(hidden_states).view(1, S_loc, nh, self.head_dim)
        if self.q_norm is not None:
            q = self.q_norm(q)
        q = q.transpose(1, 2)
        q = (q * cos_u) + (rotate_half(q) * sin_u)
        qb = gather_seq_scatter_heads_async(
            q.squeeze(0).transpose(0, 1).contiguous(), group=group)

        k = self.k_proj(hidden_states).view(1, S_loc, nkv, self.head_dim)
        if self.k_norm is not None:
            k = self.k_norm(k)
        k = k.transpose(1, 2)
        k = (k * cos_u) + (rotate_half(k) * sin_u)
        if rep > 1:
            k = torch.repeat_interleave(k, dim=1, repeats=rep)
        kb = gather_seq_scatter_heads_async(
            k.squeeze(0).transpose(0, 1).contiguous(), group=group)

        v = self.v_proj(hidden_states).view(1, S_loc, nkv, self.head_dim).transpose(1, 2)
        if rep > 1:
            v = torch.repeat_interleave(v, dim=1, repeats=rep)
        vb = gather_seq_scatter_heads_async(
            v.squeeze(0).transpose(0, 1).contiguous(), group=group)

        qg = wait_gathered(qb, group=group).transpose(0, 1)[None]  # [1, h/sp, S, D]
        kg = wait_gathered(kb, group=group).transpose(0, 1)[None]
        vg = wait_gathered(vb, group=group).transpose(0, 1)[None]
        if veomni_attention.use_non_eager_impl:
            # gathered-sequence core through the bound HIP pair (exchange
            # already done here, hence skip_ulysses)
            out, _ = veomni_attention(self, qg, kg, vg, None, dropout=0.0,
                                      scaling=self.scaling, skip_ulysses=True,
                                      **kwargs)  # [1, S, h/sp, D]
        else:
            out, _ = sdpa_attention(self, qg, kg, vg, None, dropout=0.0,
                                    scaling=self.scaling, **kwargs)
        out = out.squeeze(0)
        out = gather_heads_scatter_seq(out, head_dim=1, seq_dim=0, group=group)
        return out[None]


class MLP(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        H, I = config.hidden_size, config.intermediate_size
        self.gate_proj = nn.Linear(H, I, bias=False)
        self.up_proj = nn.Linear(H, I, bias=False)
        self.down_proj = nn.Linear(I, H, bias=False)

    def forward(self, x):
        if veomni_swiglu_mlp.use_non_eager_impl:
            return veomni_swiglu_mlp(self, x)
        return self.down_proj(F.silu(self.gate_proj(x)) * self.up_proj(x))


class TopKRouter(nn.Module):
    """Ref router semantics: raw logits; fp32 softmax; topk; renorm; cast back."""

    def __init__(self, config: ModelConfig):
        super().__init__()
        self.top_k = config.num_experts_per_tok
        self.num_experts = config.num_experts
        self.norm_topk_prob = config.norm_topk_prob
        self.hidden_dim = config.hidden_size
        self.weight = nn.Parameter(torch.zeros(self.num_experts, self.hidden_dim))

    def forward(self, hidden_states):
        hidden_states = hidden_states.reshape(-1, self.hidden_dim)
        router_logits = F.linear(hidden_states, self.weight)
        routing_weights = F.softmax(router_logits, dtype=torch.float, dim=-1)
        top_v, top_i = torch.topk(routing_weights, self.top_k, dim=-1)
        if self.norm_topk_prob:
            top_v = top_v / top_v.sum(dim=-1, keepdim=True)
        return router_logits, top_v.to(router_logits.dtype), top_i


class _EagerEPExperts:
    """Eager expert compute for the EP dispatch path (plain autograd ops;
    routing weights applied later by unpermute, matching the reference's
    post-down_proj weighting)."""

    @staticmethod
    def apply(permute_tokens, cumsum, gate_up_w, down_w):
        outs = []
        start = 0
        for g in range(gate_up_w.shape[0]):
            end = int(cumsum[g])
            cur = permute_tokens[start:end]
            gate, up = F.linear(cur, gate_up_w[g]).chunk(2, dim=-1)
            outs.append(F.linear(F.silu(gate) * up, down_w[g]))
            start = end
        return torch.cat(outs, dim=0)


class Experts(nn.Module):
    """Merged gate_up [E,2I,H] + down [E,H,I] expert bank (HF v5 layout)."""

    def __init__(self, config: ModelConfig):
        super().__init__()
        self.num_experts = config.num_experts
        self.hidden_dim = config.hidden_size
        self.intermediate_dim = config.moe_intermediate_size
        self.gate_up_proj = nn.Parameter(
            torch.empty(self.num_experts, 2 * self.intermediate_dim, self.hidden_dim)
        )
        self.down_proj = nn.Parameter(
            torch.empty(self.num_experts, self.hidden_dim, self.intermediate_dim)
        )

    def forward(self, hidden_states, top_k_index, top_k_weights):
        if veomni_moe_experts_forward.use_non_eager_impl:
            return veomni_moe_experts_forward(self, hidden_states, top_k_index, top_k_weights)
        if get_parallel_state().ep_enabled:
            # eager under EP still routes through the dispatch/combine path
            # (the reference's eager-EP equally uses the EP classes; the
            # local expert bank only holds E/ep slices, so the plain loop
            # below would index out of range)
            from ..distributed.moe import dispatch_to_ep_class

            return dispatch_to_ep_class(
                _EagerEPExperts, self.num_experts, top_k_weights, top_k_index,
                hidden_states, self.gate_up_proj, self.down_proj)
        # eager loop — weights applied AFTER down_proj (ref :266-294)
        final = torch.zeros_like(hidden_states)
        with torch.no_grad():
            expert_mask = F.one_hot(top_k_index, num_classes=self.num_experts).permute(2, 1, 0)
            expert_hit = torch.greater(expert_mask.sum(dim=(-1, -2)), 0).nonzero()
        for e_idx in expert_hit:
            e = int(e_idx[0])
            top_k_pos, token_idx = torch.where(expert_mask[e])
            cur = hidden_states[token_idx]
            gate_up = F.linear(cur, self.gate_up_proj[e])
            gate, up = gate_up.chunk(2, dim=-1)
            h = F.silu(gate) * up
            h = F.linear(h, self.down_proj[e])
            h = h * top_k_weights[token_idx, top_k_pos, None]
            final.index_add_(0, token_idx, h.to(final.dtype))
        return final


class SparseMoeBlock(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        self.experts = Experts(config)
        self.gate = TopKRouter(config)

    def forward(self, hidden_states):
        b, s, h = hidden_states.shape
        flat = hidden_states.view(-1, h)
        router_logits, routing_weights, selected = self.gate(flat)
        out = self.experts(flat, selected, routing_weights)
        return out.reshape(b, s, h), router_logits


class DecoderLayer(nn.Module):
    def __init__(self, config: ModelConfig, layer_idx: int):
        super().__init__()
        self.self_attn = Attention(config, layer_idx)
        self.mlp = SparseMoeBlock(config) if config.is_moe else MLP(config)
        self.input_layernorm = RMSNorm(config.hidden_size, eps=config.rms_norm_eps)
        self.post_attention_layernorm = RMSNorm(config.hidden_size, eps=config.rms_norm_eps)
        self.is_moe = config.is_moe

    def forward(self, hidden_states, position_embeddings, attn_kwargs=None):
        residual = hidden_states
        hidden_states = self.input_layernorm(hidden_states)
        hidden_states = self.self_attn(hidden_states, position_embeddings,
                                       **(attn_kwargs or {}))
        hidden_states = residual + hidden_states

        residual = hidden_states
        hidden_states = self.post_attention_layernorm(hidden_states)
        router_logits = None
        if self.is_moe:
            hidden_states, router_logits = self.mlp(hidden_states)
        else:
            hidden_states = self.mlp(hidden_states)
        hidden_states = residual + hidden_states
        return hidden_states, router_logits


class Model(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        self.config = config
        self.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        self.layers = nn.ModuleList(
            DecoderLayer(config, i) for i in range(config.num_hidden_layers)
        )
        self.norm = RMSNorm(config.hidden_size, eps=config.rms_norm_eps)
        self.rotary_emb = RotaryEmbedding(config)

    def forward(self, input_ids, position_ids=None, use_checkpoint=False,
                attn_kwargs=None, inputs_embeds=None):
        hidden = inputs_embeds if inputs_embeds is not None else self.embed_tokens(input_ids)
        if position_ids is None:
            position_ids = torch.arange(hidden.shape[1], device=hidden.device)[None]
        pos_emb = self.rotary_emb(hidden, position_ids)
        router_logits = []
        for layer in self.layers:
            if use_checkpoint and self.training:
                h, rl = torch.utils.checkpoint.checkpoint(
                    layer, hidden, pos_emb, attn_kwargs, use_reentrant=False
                )
            else:
                h, rl = layer(hidden, pos_emb, attn_kwargs)
            hidden = h
            if rl is not None:
                router_logits.append(rl)
        return self.norm(hidden), tuple(router_logits)


class ForCausalLM(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        self.config = config
        self.model = Model(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.use_checkpoint = False
        self.reset_parameters()

    # ------------------------------------------------------------ plumbing
    def get_decoder_layers(self):
        return list(self.model.layers)

    def get_parallel_plan(self):
        from ..distributed.parallel_plan import ParallelPlan

        # Ref models/transformers/qwen3_moe/parallel_plan.py:6-16
        return ParallelPlan(
            extra_parallel_plan={
                "ep": {
                    "model.layers.*.mlp.experts.gate_up_proj": 0,
                    "model.layers.*.mlp.experts.down_proj": 0,
                }
            }
        )

    @torch.no_grad()
    def reset_parameters(self, seed: int = 1234):
        """Seeded init (HF _init_weights pattern: normal(0, initializer_range)).

        Deterministic across ranks: one generator, sorted param order. On GPU
        the generator lives on-device (fast for multi-B-param models) — every
        rank draws the identical sequence, so replicated params agree before
        FSDP sharding."""
        std = self.config.initializer_range
        try:
            dev = next(self.parameters()).device
        except StopIteration:
            return
        if dev.type == "meta":
            return  # empty-init build (bench cpu_baseline): filled later
        g = torch.Generator(device=dev).manual_seed(seed)
        for name, p in sorted(self.named_parameters(), key=lambda kv: kv[0]):
            if name.endswith("layernorm.weight") or name.endswith("norm.weight") or ".q_norm" in name or ".k_norm" in name:
                p.fill_(1.0)
            elif name.endswith(".bias"):
                p.zero_()
            else:
                buf = torch.empty(p.shape, dtype=torch.float32, device=dev)
                buf.normal_(0.0, std, generator=g)
                p.copy_(buf.to(p.dtype))

    # ------------------------------------------------------------- forward
    def forward(self, input_ids, labels=None, position_ids=None, **kwargs):
        # packed-varlen: turn the collator's cu_seq_lens kwargs
        # (data_collator.py:50, PackingCollator) into per-token document
        # bounds ONCE per step; the attention slot applies the
        # block-diagonal causal mask (ref attention/flash.py:61-91).
        attn_kwargs = None
        cu = kwargs.get("cu_seq_lens_q")
        if cu is not None:
            from ..ops.kernels.attention import docs_from_cu_seqlens

            ds, de = docs_from_cu_seqlens(cu, int(cu.flatten()[-1]))
            if ds is not None:
                attn_kwargs = {"doc_start": ds, "doc_end": de}
        hidden, router_logits = self.model(
            input_ids, position_ids=position_ids,
            use_checkpoint=self.use_checkpoint, attn_kwargs=attn_kwargs,
            inputs_embeds=kwargs.get("inputs_embeds"),
        )
        if labels is None:
            logits = self.lm_head(hidden)
            return logits, None

        if veomni_causal_lm_loss.use_non_eager_impl:
            # ForCausalLMLoss-shaped 3-tuple contract (ref cross_entropy/
            # __init__.py:89-221): (loss, logits|None, aux|None).
            loss, _, _ = veomni_causal_lm_loss(
                hidden_states=hidden, weights=self.lm_head.weight, labels=labels
            )
        else:
            loss = self._eager_chunk_loss(hidden, labels)

        aux = None
        if self.config.is_moe and self.config.router_aux_loss_coef > 0 and router_logits:
            if veomni_load_balancing_loss.use_non_eager_impl:
                aux = veomni_load_balancing_loss(
                    router_logits, self.config.num_experts, self.config.num_experts_per_tok, None
                )
            else:
                aux = _eager_load_balancing_loss(
                    router_logits, self.config.num_experts, self.config.num_experts_per_tok
                )
            loss = loss + self.config.router_aux_loss_coef * aux.to(loss.device)
        return loss, aux

    def _eager_chunk_loss(self, hidden, labels, chunk_size: int = 1024):
        """Chunked fused-linear CE, reference-default semantics
        (ref chunk_loss.py:89-144): causal shift unless SP pre-shifted,
        per-chunk F.linear -> fp32 CE, sum/num_valid; SP loss reduce."""
        sp_enabled = get_parallel_state().sp_enabled
        orig_labels = labels
        if not sp_enabled:
            labels = labels[..., 1:].contiguous()
            hidden = hidden[..., :-1, :].contiguous()
        num_items = (labels != -100).sum()
        total = hidden.new_zeros((), dtype=torch.float32)
        for hs, lb in zip(torch.split(hidden, chunk_size, dim=1),
                          torch.split(labels, chunk_size, dim=1)):
            logits = F.linear(hs.reshape(-1, hs.shape[-1]), self.lm_head.weight).float()
            ls = F.cross_entropy(logits, lb.reshape(-1), ignore_index=-100, reduction="sum")
            total = total + ls / num_items.clamp_min(1)
        if sp_enabled:
            from ..distributed.sequence_parallel import reduce_sequence_parallel_loss

            num_valid = (orig_labels != -100).sum()
            total = reduce_sequence_parallel_loss(total, num_valid)
        return total


def _eager_load_balancing_loss(gate_logits_tuple, num_experts, top_k, attention_mask=None):
    """Switch aux loss (ref load_balancing_loss/eager.py:28-114)."""
    expert_count = torch.zeros(num_experts, dtype=torch.float32, device=gate_logits_tuple[0].device)
    prob_sum = torch.zeros_like(expert_count)
    total = torch.tensor(0.0, device=expert_count.device)
    for layer_logits in gate_logits_tuple:
        probs = torch.softmax(layer_logits.float(), dim=-1)
        _, selected = torch.topk(probs, top_k, dim=-1)
        prob_sum += probs.sum(dim=0)
        expert_count.scatter_add_(0, selected.reshape(-1),
                                  torch.ones(selected.numel(), dtype=torch.float32, device=expert_count.device))
        total = total + layer_logits.shape[0]
    return torch.dot(expert_count, prob_sum) * (num_experts / (total * total))


def bind_ops(impl: str | dict = "eager") -> None:
    """Bind every model OpSlot. `impl` may be one name for all slots or a
    per-op dict (OpsImplementationConfig analog, ref auto.py:63-107)."""
    import veomni_amd.ops  # noqa: F401  (ensures kernel registrations ran)

    slots = {
        "rms_norm": veomni_rms_norm,
        "rotary_pos_emb": veomni_apply_rotary_pos_emb,
        "swiglu_mlp": veomni_swiglu_mlp,
        "moe_experts": veomni_moe_experts_forward,
        "cross_entropy_loss": veomni_causal_lm_loss,
        "load_balancing_loss": veomni_load_balancing_loss,
        "attention": veomni_attention,
    }
    for op_name, slot in slots.items():
        name = impl if isinstance(impl, str) else impl.get(op_name, "eager")
        slot.bind(name)
