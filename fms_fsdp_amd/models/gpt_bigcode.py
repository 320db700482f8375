"""GPT-BigCode (starcoder-family) model — speculator base-model support.

Parity target: the reference's EmbedGPTBigCode speculator base
(speculator/train_speculator_utils.py:469-492 there, backed by fms
gpt_bigcode). Multi-query attention (kvheads=1), learned absolute
positions, LayerNorm + biases, gelu MLP. Attention dispatches to the
CDNA4 flash kernel; LayerNorm runs through torch's native op (this
family exists for speculator training, not the pretraining hot path).
"""

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from fms_fsdp_amd import ops


@dataclass
class GPTBigCodeConfig:
    src_vocab_size: int = 49152
    emb_dim: int = 6144
    nheads: int = 48
    nlayers: int = 40
    max_expected_seq_len: int = 8192
    hidden_grow_factor: float = 4.0
    ln_eps: float = 1e-5

    @property
    def head_dim(self):
        return self.emb_dim // self.nheads


class BigCodeAttention(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.nheads = cfg.nheads
        self.head_dim = cfg.head_dim
        # multi-query: one shared kv head
        self.qkv = nn.Linear(cfg.emb_dim, (cfg.nheads + 2) * cfg.head_dim)
        self.proj = nn.Linear(cfg.emb_dim, cfg.emb_dim)

    def reset_parameters(self):
        for lin in (self.qkv, self.proj):
            nn.init.trunc_normal_(lin.weight, std=0.02)
            nn.init.zeros_(lin.bias)

    def forward(self, x, cache=None):
        b, s, _ = x.shape
        q, k, v = self.qkv(x).split(
            [self.nheads * self.head_dim, self.head_dim, self.head_dim], -1)
        q = q.view(b, s, self.nheads, self.head_dim)
        k = k.view(b, s, 1, self.head_dim)
        v = v.view(b, s, 1, self.head_dim)
        if cache is not None:
            if cache.get("k") is not None:
                k = torch.cat([cache["k"], k], dim=1)
                v = torch.cat([cache["v"], v], dim=1)
            cache["k"], cache["v"] = k, v
            o = F.scaled_dot_product_attention(
                q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
                is_causal=(s == k.shape[1]), enable_gqa=True).transpose(1, 2)
        else:
            o = ops.attention_causal(q.contiguous(), k.contiguous(),
                                     v.contiguous())
        return self.proj(o.reshape(b, s, -1))


class BigCodeBlock(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.ln1 = nn.LayerNorm(cfg.emb_dim, eps=cfg.ln_eps)
        self.attn = BigCodeAttention(cfg)
        self.ln2 = nn.LayerNorm(cfg.emb_dim, eps=cfg.ln_eps)
        hidden = int(cfg.emb_dim * cfg.hidden_grow_factor)
        self.fc = nn.Linear(cfg.emb_dim, hidden)
        self.out = nn.Linear(hidden, cfg.emb_dim)

    def reset_parameters(self):
        self.ln1.reset_parameters()
        self.ln2.reset_parameters()
        self.attn.reset_parameters()
        for lin in (self.fc, self.out):
            nn.init.trunc_normal_(lin.weight, std=0.02)
            nn.init.zeros_(lin.bias)

    def forward(self, x, cache=None):
        x = x + self.attn(self.ln1(x), cache)
        x = x + self.out(F.gelu(self.fc(self.ln2(x)), approximate="tanh"))
        return x


class GPTBigCode(nn.Module):
    def __init__(self, cfg: GPTBigCodeConfig):
        super().__init__()
        self.config = cfg
        self.embedding = nn.Embedding(cfg.src_vocab_size, cfg.emb_dim)
        self.wpe = nn.Embedding(cfg.max_expected_seq_len, cfg.emb_dim)
        self.layers = nn.ModuleList(
            [BigCodeBlock(cfg) for _ in range(cfg.nlayers)])
        self.ln_f = nn.LayerNorm(cfg.emb_dim, eps=cfg.ln_eps)
        self.lm_head = nn.Linear(cfg.emb_dim, cfg.src_vocab_size, bias=False)

    def reset_parameters(self):
        nn.init.trunc_normal_(self.embedding.weight, std=0.02)
        nn.init.trunc_normal_(self.wpe.weight, std=0.02)
        nn.init.trunc_normal_(self.lm_head.weight, std=0.02)
        self.ln_f.reset_parameters()
        for l in self.layers:
            l.reset_parameters()

    def forward(self, tokens, labels=None, include_embeds=False, pos0=0,
                caches=None):
        b, s = tokens.shape
        pos = torch.arange(pos0, pos0 + s, device=tokens.device)
        x = self.embedding(tokens) + self.wpe(pos)
        for i, layer in enumerate(self.layers):
            x = layer(x, caches[i] if caches is not None else None)
        x = self.ln_f(x)
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
            logits, hid = self.forward(cur, include_embeds=True, pos0=pos,
                                       caches=caches)
            pos += cur.shape[1]
            last = logits[:, -1]
            if do_sample:
                nxt = torch.multinomial(
                    torch.softmax(last.float() / temperature, -1), 1)
            else:
                nxt = last.argmax(-1, keepdim=True)
            embeds.append(hid[:, -1:])
            tokens = torch.cat([tokens, nxt], dim=1)
            cur = nxt
        if include_embeds:
            return tokens, torch.cat(embeds, dim=1)
        return tokens

    def param_count(self):
        return sum(p.numel() for p in self.parameters())
