"""MLPSpeculator: n-lookahead speculative-decoding heads.

Replaces the external `fms_extras` MLPSpeculator the reference trains
(speculator/train_speculator.py:8-11 there). Head i consumes the running
state and the previous predicted/ground-truth token embedding:
    state_i = act(LN_i(W_i state_{i-1} * sw + E_i(tok_{i-1}) * ew))
    logits_i = head_i(state_i)
with sw/ew the state/emb mixing weights; tie_weights shares E/LN/head/W
across heads (except the first projection, whose input dim differs).
"""

import math

import torch
import torch.nn as nn


class LayerNormParameterized(nn.Module):
    def __init__(self, dim, elementwise_scale=True, elementwise_shift=True,
                 eps=1e-6):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim)) if elementwise_scale else None
        self.bias = nn.Parameter(torch.zeros(dim)) if elementwise_shift else None

    def forward(self, x):
        xf = x.float()
        y = (xf - xf.mean(-1, keepdim=True)) * torch.rsqrt(
            xf.var(-1, keepdim=True, unbiased=False) + self.eps)
        if self.weight is not None:
            y = y * self.weight.float()
        if self.bias is not None:
            y = y + self.bias.float()
        return y.to(x.dtype)

    def reset_parameters(self):
        if self.weight is not None:
            nn.init.ones_(self.weight)
        if self.bias is not None:
            nn.init.zeros_(self.bias)


class MLPSpeculator(nn.Module):
    def __init__(self, emb_dim, inner_dim, vocab_size, n_predict,
                 tie_weights=True, scale_input=True):
        super().__init__()
        self.emb_dim = emb_dim
        self.inner_dim = inner_dim if inner_dim != 0 else emb_dim
        self.vocab_size = vocab_size
        self.n_predict = n_predict
        self.tie_weights = tie_weights
        self.scale_input = scale_input

        self.state_weight = 0.5 ** (0.5 / n_predict)
        self.emb_weight = math.sqrt(1 - self.state_weight ** 2)
        self.activation = nn.GELU()

        n_emb = 1 if tie_weights else n_predict
        n_proj = 2 if tie_weights and n_predict > 1 else n_predict
        self.emb = nn.ModuleList(
            [nn.Embedding(vocab_size, self.inner_dim) for _ in range(n_emb)])
        self.proj = nn.ModuleList([
            nn.Linear(emb_dim if i == 0 else self.inner_dim, self.inner_dim,
                      bias=False)
            for i in range(min(n_proj, n_predict))])
        self.head = nn.ModuleList(
            [nn.Linear(self.inner_dim, vocab_size, bias=False)
             for _ in range(n_emb)])
        self.ln = nn.ModuleList(
            [LayerNormParameterized(self.inner_dim) for _ in range(n_emb)])
        self.ln_input = LayerNormParameterized(emb_dim) if scale_input else None

    def _pick(self, mods, i):
        return mods[min(i, len(mods) - 1)]

    def _proj_for(self, i):
        # head 0 has a distinct input dim; with tied weights heads >=1 share
        return self.proj[0] if i == 0 else self._pick(self.proj, 1) \
            if self.tie_weights else self.proj[i]

    def reset_parameters(self):
        for m in list(self.emb) + list(self.head) + list(self.proj):
            nn.init.trunc_normal_(m.weight, std=0.02)
        for m in self.ln:
            m.reset_parameters()
        if self.ln_input is not None:
            self.ln_input.reset_parameters()

    def forward(self, state, inds):
        """state (b, s, emb_dim): base-model hidden states;
        inds (b, s + n_predict - 1): token ids aligned so head i at
        position t predicts inds[t + i + 1].
        Returns logits (n_predict, b, s, vocab)."""
        b, s, _ = state.shape
        if self.ln_input is not None:
            state = self.ln_input(state) / math.sqrt(2)
        out = []
        for i in range(self.n_predict):
            z = self._pick(self.emb, i)(inds[:, i:i + s])
            state = self._proj_for(i)(state) * self.state_weight \
                + z * self.emb_weight
            state = self.activation(self._pick(self.ln, i)(state))
            out.append(self._pick(self.head, i)(state))
        return torch.stack(out, dim=0)

    def param_count(self):
        return sum(p.numel() for p in self.parameters())
