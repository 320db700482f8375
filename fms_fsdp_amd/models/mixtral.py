"""Mixtral (MoE Llama) model — speculator base-model support.

Parity target: the reference's EmbedMixtral speculator base
(speculator/train_speculator_utils.py:495-523 there, dense-executed fms
mixtral; expert-parallel routing is out of reference scope, SURVEY.md
§2.2 EP row). Llama backbone (RMSNorm, rotary GQA via the CDNA4 kernels)
with a top-2-of-8 sparse-MoE MLP executed by expert-gathered batching.
"""

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from fms_fsdp_amd import ops
from fms_fsdp_amd.models.llama import Attention, RMSNorm, RotaryEmbedding


@dataclass
class MixtralConfig:
    src_vocab_size: int = 32000
    emb_dim: int = 4096
    nheads: int = 32
    kvheads: int = 8
    nlayers: int = 32
    hidden_dim: int = 14336
    num_experts: int = 8
    top_k: int = 2
    max_expected_seq_len: int = 32768
    rope_theta: float = 1000000.0
    norm_eps: float = 1e-5

    @property
    def head_dim(self):
        return self.emb_dim // self.nheads


class MoEMLP(nn.Module):
    """Top-k routed SwiGLU experts, executed densely by expert-gathered
    batching (every expert's tokens run as one GEMM batch)."""

    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.num_experts = cfg.num_experts
        self.top_k = cfg.top_k
        self.router = nn.Linear(cfg.emb_dim, cfg.num_experts, bias=False)
        self.wg1 = nn.ModuleList(
            [nn.Linear(cfg.emb_dim, 2 * cfg.hidden_dim, bias=False)
             for _ in range(cfg.num_experts)])
        self.w2 = nn.ModuleList(
            [nn.Linear(cfg.hidden_dim, cfg.emb_dim, bias=False)
             for _ in range(cfg.num_experts)])

    def reset_parameters(self):
        nn.init.trunc_normal_(self.router.weight, std=0.02)
        for lin in list(self.wg1) + list(self.w2):
            nn.init.trunc_normal_(lin.weight, std=0.02)

    def forward(self, x):
        b, s, e = x.shape
        flat = x.view(-1, e)
        logits = self.router(flat).float()
        weights, experts = torch.topk(torch.softmax(logits, -1), self.top_k, -1)
        weights = weights / weights.sum(-1, keepdim=True)
        out = torch.zeros_like(flat)
        for ex in range(self.num_experts):
            tok, slot = torch.where(experts == ex)
            if tok.numel() == 0:
                continue
            h = ops.swiglu(self.wg1[ex](flat[tok]))
            out.index_add_(0, tok,
                           (self.w2[ex](h).float()
                            * weights[tok, slot, None]).to(out.dtype))
        return out.view(b, s, e)


class MixtralBlock(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        from fms_fsdp_amd.models.llama import LlamaConfig
        acfg = LlamaConfig(src_vocab_size=cfg.src_vocab_size,
                           emb_dim=cfg.emb_dim, nheads=cfg.nheads,
                           kvheads=cfg.kvheads, nlayers=cfg.nlayers,
                           max_expected_seq_len=cfg.max_expected_seq_len,
                           rope_theta=cfg.rope_theta, norm_eps=cfg.norm_eps)
        self.attn_norm = RMSNorm(cfg.emb_dim, cfg.norm_eps)
        self.attn = Attention(acfg)
        self.moe_norm = RMSNorm(cfg.emb_dim, cfg.norm_eps)
        self.moe = MoEMLP(cfg)

    def reset_parameters(self):
        self.attn_norm.reset_parameters()
        self.attn.reset_parameters()
        self.moe_norm.reset_parameters()
        self.moe.reset_parameters()

    def forward(self, x, cos, sin, cache=None):
        x = x + self.attn(self.attn_norm(x), cos, sin, cache)
        x = x + self.moe(self.moe_norm(x))
        return x


class Mixtral(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.config = cfg
        self.embedding = nn.Embedding(cfg.src_vocab_size, cfg.emb_dim)
        self.rot_emb = RotaryEmbedding(cfg.head_dim, cfg.max_expected_seq_len,
                                       cfg.rope_theta)
        self.layers = nn.ModuleList(
            [MixtralBlock(cfg) for _ in range(cfg.nlayers)])
        self.norm = RMSNorm(cfg.emb_dim, cfg.norm_eps)
        self.lm_head = nn.Linear(cfg.emb_dim, cfg.src_vocab_size, bias=False)

    def reset_parameters(self):
        nn.init.trunc_normal_(self.embedding.weight, std=0.02)
        nn.init.trunc_normal_(self.lm_head.weight, std=0.02)
        self.norm.reset_parameters()
        for l in self.layers:
            l.reset_parameters()

    def forward(self, tokens, labels=None, include_embeds=False):
        b, s = tokens.shape
        x = self.embedding(tokens)
        cos, sin = self.rot_emb.get(s, x.device)
        for layer in self.layers:
            x = layer(x, cos, sin)
        x = self.norm(x)
        if labels is not None:
            loss = ops.linear_cross_entropy(x, self.lm_head.weight, labels)
            return (loss, x) if include_embeds else loss
        logits = self.lm_head(x)
        return (logits, x) if include_embeds else logits

    @torch.no_grad()
    def generate(self, input_ids, max_new_tokens, temperature=1.0,
                 do_sample=True, include_embeds=False):
        caches = [{} for _ in self.layers]
        tokens = input_ids
        embeds = []
        cur = input_ids
        pos = 0
        for _ in range(max_new_tokens):
            x = self.embedding(cur)
            total = pos + cur.shape[1]
            cos, sin = self.rot_emb.get(total, x.device)
            cos_c, sin_c = cos[pos:total], sin[pos:total]
            for layer, cache in zip(self.layers, caches):
                x = layer(x, cos_c, sin_c, cache)
            x = self.norm(x)
            last = self.lm_head(x[:, -1:])[:, -1]
            if do_sample:
                nxt = torch.multinomial(
                    torch.softmax(last.float() / temperature, -1), 1)
            else:
                nxt = last.argmax(-1, keepdim=True)
            embeds.append(x[:, -1:])
            tokens = torch.cat([tokens, nxt], dim=1)
            pos = total
            cur = nxt
        if include_embeds:
            return tokens, torch.cat(embeds, dim=1)
        return tokens

    def param_count(self):
        return sum(p.numel() for p in self.parameters())
