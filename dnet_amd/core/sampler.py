"""Token sampling: temperature / top-k / top-p / min-p + logprobs.

Reference counterpart: src/dnet/core/decoding/sampler.py (mlx_lm
make_sampler); here implemented on torch logits [B, V].
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch


@dataclass
class DecodingConfig:
    temperature: float = 0.0
    top_k: int = 0
    top_p: float = 1.0
    min_p: float = 0.0
    logprobs: bool = False
    top_logprobs: int = 0


class Sampler:
    def __init__(self, cfg: DecodingConfig | None = None,
                 generator: Optional[torch.Generator] = None):
        self.cfg = cfg or DecodingConfig()
        self.generator = generator

    def sample(self, logits: torch.Tensor):
        """logits [B, V] -> (tokens [B] int64, logprob [B] or None,
        top_logprobs list or None)."""
        c = self.cfg
        lf = logits.float()
        if c.temperature <= 0.0:
            tok = lf.argmax(dim=-1)
        else:
            x = lf / c.temperature
            if c.top_k and c.top_k > 0:
                kth = torch.topk(x, min(c.top_k, x.shape[-1]), dim=-1).values[..., -1:]
                x = x.masked_fill(x < kth, float("-inf"))
            if c.top_p < 1.0:
                sx, si = torch.sort(x, descending=True, dim=-1)
                probs = torch.softmax(sx, dim=-1)
                cum = probs.cumsum(dim=-1)
                keep = cum - probs < c.top_p  # keep at least the top token
                sx = sx.masked_fill(~keep, float("-inf"))
                x = torch.full_like(x, float("-inf")).scatter_(-1, si, sx)
            if c.min_p > 0.0:
                p = torch.softmax(x, dim=-1)
                x = x.masked_fill(p < c.min_p * p.amax(-1, keepdim=True),
                                  float("-inf"))
            probs = torch.softmax(x, dim=-1)
            # degenerate rows (NaN logits, or everything masked) would trip
            # multinomial's device-side assert and abort the whole process
            # (HSA exception) — fall back to greedy for those rows, without
            # a host sync (stays usable inside the serving loop)
            probs = torch.nan_to_num(probs, nan=0.0, posinf=0.0)
            ok = probs.sum(dim=-1, keepdim=True) > 0
            fallback = torch.nn.functional.one_hot(
                lf.nan_to_num(nan=0.0).argmax(dim=-1), lf.shape[-1]).float()
            probs = torch.where(ok, probs, fallback)
            tok = torch.multinomial(probs, 1, generator=self.generator).squeeze(-1)
        if not c.logprobs:
            return tok, None, None
        logp = lf - torch.logsumexp(lf, dim=-1, keepdim=True)
        chosen = logp.gather(-1, tok.unsqueeze(-1)).squeeze(-1)
        tops = None
        if c.top_logprobs:
            tv, ti = torch.topk(logp, c.top_logprobs, dim=-1)
            tops = [{int(i): float(v) for v, i in zip(tv[b], ti[b])}
                    for b in range(logits.shape[0])]
        return tok, chosen, tops


class RowSampler:
    """Per-row sampling for slot-batched serving: each row of the logits
    batch carries its own request's decoding params (temperature / top_k /
    top_p / min_p), vectorized in one pass. Greedy rows (temperature 0)
    take the argmax; sampled rows go through per-row filtered multinomial
    with the same degenerate-row guard as ``Sampler``."""

    def __init__(self, batch: int, device=None):
        self.batch = batch
        self.device = device
        self.temp = torch.zeros(batch, device=device)
        self.top_p = torch.ones(batch, device=device)
        self.top_k = torch.zeros(batch, dtype=torch.long, device=device)
        self.min_p = torch.zeros(batch, device=device)
        self._has_min_p = False    # python-side (no device sync per step)
        # per-row RNG for seeded requests (OpenAI `seed`); None = shared
        # default stream. Seeded rows sample via inverse-CDF with a u drawn
        # from their own generator, so one request's stream is reproducible
        # regardless of which other slots are active.
        self.gens: list = [None] * batch
        # per-row logprob config (OpenAI `logprobs`/`top_logprobs`); the
        # sampler computes them for the WHOLE batch when any row wants
        # them, and the emitter picks per row
        self.want_lp: list = [False] * batch
        self.n_top: list = [0] * batch
        self.last_logp = None    # [B] device, chosen-token logprob
        self.last_topv = None    # [B, K] device
        self.last_topi = None
        # python-side mirrors: the hot sample() picks its fast path
        # (greedy-only / filterless) without device syncs
        self._temp_py = [0.0] * batch
        self._filt_py = [False] * batch

    def set_row(self, i: int, cfg: DecodingConfig, seed=None):
        self.temp[i] = cfg.temperature
        self.top_p[i] = cfg.top_p
        self.top_k[i] = cfg.top_k
        self.min_p[i] = cfg.min_p
        if cfg.min_p > 0:
            self._has_min_p = True
        self.want_lp[i] = bool(cfg.logprobs)
        self.n_top[i] = int(cfg.top_logprobs or 0)
        self._temp_py[i] = float(cfg.temperature)
        self._filt_py[i] = bool(cfg.top_k or cfg.top_p < 1.0 or cfg.min_p)
        g = None
        if seed is not None:
            g = torch.Generator(device=self.device)
            g.manual_seed(int(seed))
        self.gens[i] = g
        return g

    def clear_row(self, i: int):
        """Reset a freed slot so stale sampling params don't keep the
        batch off the greedy fast path."""
        self.temp[i] = 0.0
        self.top_p[i] = 1.0
        self.top_k[i] = 0
        self.min_p[i] = 0.0
        self._temp_py[i] = 0.0
        self._filt_py[i] = False
        self.want_lp[i] = False
        self.n_top[i] = 0
        self.gens[i] = None

    def _logprobs_for(self, lf, tok):
        B = lf.shape[0]
        if any(self.want_lp[:B]):
            logp = lf - torch.logsumexp(lf, dim=-1, keepdim=True)
            self.last_logp = logp.gather(-1, tok.unsqueeze(-1)).squeeze(-1)
            kmax = max(self.n_top[:B])
            if kmax > 0:
                self.last_topv, self.last_topi = torch.topk(logp, kmax,
                                                            dim=-1)
            else:
                self.last_topv = self.last_topi = None
        else:
            self.last_logp = self.last_topv = self.last_topi = None

    def sample(self, logits: torch.Tensor) -> torch.Tensor:
        lf = logits.float()
        B, V = lf.shape
        greedy_tok = lf.argmax(dim=-1)
        # fast path: every row greedy (the common serving batch) — the
        # full sort/softmax/multinomial over [B, V] cost ~25 ms/step on
        # a 152k vocab and collapsed serving throughput 4x
        if not any(t > 0 for t in self._temp_py[:B]):
            self._logprobs_for(lf, greedy_tok)
            return greedy_tok
        t = self.temp.clamp_min(1e-6).unsqueeze(-1)
        x = lf / t
        if not any(self._filt_py[:B]):
            # sampled but filterless: plain softmax + draw (no sort)
            probs = torch.softmax(x, dim=-1)
            probs = torch.nan_to_num(probs, nan=0.0, posinf=0.0)
            ok = probs.sum(dim=-1, keepdim=True) > 0
            fallback = torch.nn.functional.one_hot(
                lf.nan_to_num(nan=0.0).argmax(dim=-1), V).float()
            probs = torch.where(ok, probs, fallback)
            tok = self._draw(probs, B, V, lf.device)
            tok = torch.where(self.temp <= 0.0, greedy_tok, tok)
            self._logprobs_for(lf, tok)
            return tok
        sx, si = torch.sort(x, descending=True, dim=-1)
        ranks = torch.arange(V, device=lf.device).expand(B, V)
        keep = torch.ones_like(sx, dtype=torch.bool)
        # per-row top-k (0 = off)
        k = torch.where(self.top_k > 0, self.top_k,
                        torch.full_like(self.top_k, V))
        keep &= ranks < k.unsqueeze(-1)
        # per-row top-p (keep at least the top token)
        probs_s = torch.softmax(sx, dim=-1)
        cum = probs_s.cumsum(dim=-1)
        keep &= (cum - probs_s) < self.top_p.unsqueeze(-1)
        sx = sx.masked_fill(~keep, float("-inf"))
        x = torch.full_like(x, float("-inf")).scatter_(-1, si, sx)
        if self._has_min_p:
            p = torch.softmax(x, dim=-1)
            thresh = self.min_p.unsqueeze(-1) * p.amax(-1, keepdim=True)
            x = x.masked_fill(p < thresh, float("-inf"))
        probs = torch.softmax(x, dim=-1)
        probs = torch.nan_to_num(probs, nan=0.0, posinf=0.0)
        ok = probs.sum(dim=-1, keepdim=True) > 0
        fallback = torch.nn.functional.one_hot(
            lf.nan_to_num(nan=0.0).argmax(dim=-1), V).float()
        probs = torch.where(ok, probs, fallback)
        tok = self._draw(probs, B, V, lf.device)
        tok = torch.where(self.temp <= 0.0, greedy_tok, tok)
        self._logprobs_for(lf, tok)
        return tok

    def _draw(self, probs, B, V, device):
        if any(g is not None for g in self.gens[:B]):
            # inverse-CDF with per-row uniforms: seeded rows draw from
            # their own generator, unseeded rows from the default stream
            u = torch.rand(B, 1, device=device)
            for i, g in enumerate(self.gens[:B]):
                if g is not None:
                    u[i] = torch.rand(1, 1, device=device, generator=g)
            cdf = probs.cumsum(-1)
            return torch.searchsorted(
                cdf, u * cdf[..., -1:]).squeeze(-1).clamp_(0, V - 1)
        return torch.multinomial(probs, 1).squeeze(-1)
