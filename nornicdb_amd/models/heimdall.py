"""Heimdall generative model: Qwen2-style causal decoder, MI355X-native.

Replaces the reference's llama.cpp Qwen-2.5-0.5B GGUF generation path
(reference pkg/localllm/llama.go:195-240, pkg/heimdall/generator_cgo.go)
with a from-scratch PyTorch-ROCm decoder: RMSNorm, RoPE, GQA attention
with KV cache, SwiGLU MLP. Weights are random-init in this offline
environment (a state_dict can be loaded when available); the contract is
architecture + decode-throughput parity.
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F


@dataclass
class HeimdallConfig:
    # Qwen-2.5-0.5B shape
    vocab_size: int = 151_936
    hidden_size: int = 896
    num_layers: int = 24
    num_heads: int = 14
    num_kv_heads: int = 2
    intermediate_size: int = 4864
    max_position: int = 4096
    rope_theta: float = 1_000_000.0
    rms_eps: float = 1e-6

    @classmethod
    def tiny(cls, **kw):
        d = dict(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, intermediate_size=128, max_position=256)
        d.update(kw)
        return cls(**d)


def _rope(x, cos, sin):
    # x: [B, H, S, D]
    d = x.shape[-1]
    x1, x2 = x[..., : d // 2], x[..., d // 2:]
    rot = torch.cat([-x2, x1], dim=-1)
    return x * cos + rot * sin


class _DecoderLayer(nn.Module):
    def __init__(self, cfg: HeimdallConfig):
        super().__init__()
        h = cfg.hidden_size
        self.hd = h // cfg.num_heads
        self.nh = cfg.num_heads
        self.nkv = cfg.num_kv_heads
        self.q_proj = nn.Linear(h, self.nh * self.hd, bias=True)
        self.k_proj = nn.Linear(h, self.nkv * self.hd, bias=True)
        self.v_proj = nn.Linear(h, self.nkv * self.hd, bias=True)
        self.o_proj = nn.Linear(self.nh * self.hd, h, bias=False)
        self.gate_proj = nn.Linear(h, cfg.intermediate_size, bias=False)
        self.up_proj = nn.Linear(h, cfg.intermediate_size, bias=False)
        self.down_proj = nn.Linear(cfg.intermediate_size, h, bias=False)
        self.ln1 = nn.RMSNorm(h, eps=cfg.rms_eps) if hasattr(nn, "RMSNorm") \
            else nn.LayerNorm(h, eps=cfg.rms_eps)
        self.ln2 = nn.RMSNorm(h, eps=cfg.rms_eps) if hasattr(nn, "RMSNorm") \
            else nn.LayerNorm(h, eps=cfg.rms_eps)

    def forward(self, x, cos, sin, kv_cache=None, pos0: int = 0):
        b, s, h = x.shape
        r = self.ln1(x)
        q = self.q_proj(r).view(b, s, self.nh, self.hd).transpose(1, 2)
        k = self.k_proj(r).view(b, s, self.nkv, self.hd).transpose(1, 2)
        v = self.v_proj(r).view(b, s, self.nkv, self.hd).transpose(1, 2)
        q = _rope(q, cos, sin)
        k = _rope(k, cos, sin)
        if kv_cache is not None:
            pk, pv = kv_cache
            if pk is not None:
                k = torch.cat([pk, k], dim=2)
                v = torch.cat([pv, v], dim=2)
            new_cache = (k, v)
        else:
            new_cache = None
        rep = self.nh // self.nkv
        ke = k.repeat_interleave(rep, dim=1)
        ve = v.repeat_interleave(rep, dim=1)
        causal = kv_cache is None or kv_cache[0] is None
        a = F.scaled_dot_product_attention(q, ke, ve, is_causal=causal and s > 1)
        a = a.transpose(1, 2).reshape(b, s, self.nh * self.hd)
        x = x + self.o_proj(a)
        r = self.ln2(x)
        x = x + self.down_proj(F.silu(self.gate_proj(r)) * self.up_proj(r))
        return x, new_cache


class HeimdallModel(nn.Module):
    def __init__(self, cfg: HeimdallConfig = None):
        super().__init__()
        self.cfg = cfg or HeimdallConfig()
        c = self.cfg
        self.embed = nn.Embedding(c.vocab_size, c.hidden_size)
        self.layers = nn.ModuleList(_DecoderLayer(c) for _ in range(c.num_layers))
        self.norm = nn.RMSNorm(c.hidden_size, eps=c.rms_eps) \
            if hasattr(nn, "RMSNorm") else nn.LayerNorm(c.hidden_size)
        self.lm_head = nn.Linear(c.hidden_size, c.vocab_size, bias=False)
        self.lm_head.weight = self.embed.weight  # tied
        # rope tables
        hd = c.hidden_size // c.num_heads
        inv = 1.0 / (c.rope_theta ** (torch.arange(0, hd, 2).float() / hd))
        t = torch.arange(c.max_position).float()
        freqs = torch.outer(t, inv)
        emb = torch.cat([freqs, freqs], dim=-1)
        self.register_buffer("rope_cos", emb.cos(), persistent=False)
        self.register_buffer("rope_sin", emb.sin(), persistent=False)

    @torch.no_grad()
    def init_small(self, std=0.02):
        for p in self.parameters():
            if p.dim() > 1:
                p.normal_(0, std)
            else:
                p.zero_()
        for m in self.modules():
            if m.__class__.__name__ in ("RMSNorm", "LayerNorm"):
                if hasattr(m, "weight") and m.weight is not None:
                    m.weight.fill_(1.0)
        return self

    def forward(self, token_ids, kv_caches=None, pos0: int = 0):
        b, s = token_ids.shape
        x = self.embed(token_ids)
        cos = self.rope_cos[pos0:pos0 + s].to(x.dtype)[None, None]
        sin = self.rope_sin[pos0:pos0 + s].to(x.dtype)[None, None]
        new_caches = []
        for i, layer in enumerate(self.layers):
            cache = kv_caches[i] if kv_caches is not None else None
            x, nc = layer(x, cos, sin, kv_cache=cache, pos0=pos0)
            new_caches.append(nc)
        x = self.norm(x)
        logits = self.lm_head(x)
        return logits, (new_caches if kv_caches is not None else None)

    @torch.no_grad()
    def generate(self, token_ids: torch.Tensor, max_new_tokens: int = 32,
                 temperature: float = 0.8, top_k: int = 40,
                 eos_id: Optional[int] = None, stream_cb=None) -> List[int]:
        """Greedy/top-k sampled decode with KV cache. Returns new token ids."""
        self.eval()
        b, s = token_ids.shape
        assert b == 1, "generate() is single-sequence"
        caches = [(None, None)] * len(self.layers)
        logits, caches = self.forward(token_ids, kv_caches=caches, pos0=0)
        out = []
        cur = None
        pos = s
        for _ in range(max_new_tokens):
            last = logits[:, -1, :].float()
            if temperature <= 0:
                nxt = int(last.argmax(-1))
            else:
                last = last / temperature
                if top_k:
                    v, ix = torch.topk(last, min(top_k, last.shape[-1]))
                    probs = torch.softmax(v, dim=-1)
                    nxt = int(ix[0, int(torch.multinomial(probs[0], 1))])
                else:
                    probs = torch.softmax(last, dim=-1)
                    nxt = int(torch.multinomial(probs[0], 1))
            out.append(nxt)
            if stream_cb:
                stream_cb(nxt)
            if eos_id is not None and nxt == eos_id:
                break
            cur = torch.tensor([[nxt]], device=token_ids.device)
            logits, caches = self.forward(cur, kv_caches=caches, pos0=pos)
            pos += 1
        return out


class GraphedDecoder:
    """hipGraph-captured single-token decode step.

    The eager decode loop launches ~150 kernels per token and is launch
    bound on small models; capturing one step as a HIP graph replays the
    whole token in a single launch. Static KV cache (attention over
    max_len with a position mask keeps every shape static), static
    input/position/output buffers.
    """

    def __init__(self, model: HeimdallModel, max_len: int = 512):
        self.m = model
        self.cfg = model.cfg
        self.max_len = min(max_len, self.cfg.max_position)
        dev = next(model.parameters()).device
        dtype = next(model.parameters()).dtype
        c = self.cfg
        hd = c.hidden_size // c.num_heads
        self.hd = hd
        self.cache_k = [torch.zeros(1, c.num_kv_heads, self.max_len, hd,
                                    device=dev, dtype=dtype)
                        for _ in range(c.num_layers)]
        self.cache_v = [torch.zeros_like(self.cache_k[0])
                        for _ in range(c.num_layers)]
        self.tok = torch.zeros(1, 1, device=dev, dtype=torch.long)
        self.pos = torch.zeros(1, device=dev, dtype=torch.long)
        self.arange = torch.arange(self.max_len, device=dev)
        self.graph = None
        self.out = None

    def _step(self):
        m, c = self.m, self.cfg
        x = m.embed(self.tok)                              # [1,1,h]
        cos = m.rope_cos.index_select(0, self.pos).to(x.dtype)[None, None]
        sin = m.rope_sin.index_select(0, self.pos).to(x.dtype)[None, None]
        keep = (self.arange <= self.pos).view(1, 1, 1, self.max_len)
        rep = c.num_heads // c.num_kv_heads
        for li, layer in enumerate(m.layers):
            r = layer.ln1(x)
            q = layer.q_proj(r).view(1, 1, layer.nh, layer.hd).transpose(1, 2)
            k = layer.k_proj(r).view(1, 1, layer.nkv, layer.hd).transpose(1, 2)
            v = layer.v_proj(r).view(1, 1, layer.nkv, layer.hd).transpose(1, 2)
            q = _rope(q, cos, sin)
            k = _rope(k, cos, sin)
            self.cache_k[li].index_copy_(2, self.pos, k)
            self.cache_v[li].index_copy_(2, self.pos, v)
            ke = self.cache_k[li].repeat_interleave(rep, dim=1)
            ve = self.cache_v[li].repeat_interleave(rep, dim=1)
            a = F.scaled_dot_product_attention(q, ke, ve, attn_mask=keep)
            a = a.transpose(1, 2).reshape(1, 1, layer.nh * layer.hd)
            x = x + layer.o_proj(a)
            r = layer.ln2(x)
            x = x + layer.down_proj(F.silu(layer.gate_proj(r)) * layer.up_proj(r))
        return m.lm_head(m.norm(x))[:, -1, :]

    def capture(self):
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                self._step()
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self.out = self._step()
        self.graph = g
        return self

    @torch.no_grad()
    def generate(self, token_ids: torch.Tensor, max_new_tokens: int = 32,
                 temperature: float = 0.8, top_k: int = 40,
                 eos_id=None, stream_cb=None):
        """Prefill with the dynamic forward, then graph-replay per token."""
        assert self.graph is not None, "call capture() first"
        m = self.m
        s = token_ids.shape[1]
        caches = [(None, None)] * len(m.layers)
        logits, caches = m.forward(token_ids, kv_caches=caches, pos0=0)
        for li, (k, v) in enumerate(caches):
            self.cache_k[li][:, :, :s] = k
            self.cache_v[li][:, :, :s] = v
        # sampling stays ON DEVICE: no host sync per token unless the
        # caller streams or uses eos early-exit
        need_sync = stream_cb is not None or eos_id is not None
        out_t = []
        last = logits[:, -1, :].float()
        pos = s

        def pick(lg):
            if temperature <= 0:
                return lg.argmax(-1)                      # [1]
            sc = lg / temperature
            if top_k:
                vv, ix = torch.topk(sc, min(top_k, sc.shape[-1]))
                j = torch.multinomial(torch.softmax(vv, -1)[0], 1)
                return ix[0].index_select(0, j)           # [1]
            return torch.multinomial(torch.softmax(sc, -1)[0], 1)

        for _ in range(max_new_tokens):
            nxt = pick(last)
            out_t.append(nxt)
            if need_sync:
                nv = int(nxt)
                if stream_cb:
                    stream_cb(nv)
                if eos_id is not None and nv == eos_id:
                    break
            if pos >= self.max_len:
                break
            self.tok.copy_(nxt.view(1, 1))
            self.pos.fill_(pos)
            self.graph.replay()
            last = self.out.float()
            pos += 1
        return [int(t) for t in torch.cat(out_t).tolist()] if out_t else []


class FusedDecoder:
    """Single-launch cooperative decode step (csrc/decode_fused.hip).

    All decoder layers for one token run in ONE hipLaunchCooperativeKernel
    (grid-wide syncs between GEMV/attention stages); only the final norm,
    lm_head GEMV and sampling remain as torch ops (~4 launches/token vs
    ~180 for the eager loop). Falls back is handled by HeimdallManager.
    Requires head_dim 64 and max_len <= 4096.
    """

    def __init__(self, model: HeimdallModel, max_len: int = 4096):
        from ..ops import require_native
        self.nat = require_native()
        self.m = model
        c = model.cfg
        self.cfg = c
        dev = next(model.parameters()).device
        assert dev.type == "cuda", "FusedDecoder is GPU-only"
        self.dev = dev
        self.hd = c.hidden_size // c.num_heads
        assert self.hd == 64, "fused decode supports head_dim 64"
        self.max_len = min(max_len, c.max_position, 4096)
        dt = next(model.parameters()).dtype
        assert dt == torch.bfloat16, "fused decode expects bf16 weights"
        self.cache_k = [torch.zeros(c.num_kv_heads, self.max_len, self.hd,
                                    device=dev, dtype=dt)
                        for _ in range(c.num_layers)]
        self.cache_v = [torch.zeros_like(self.cache_k[0])
                        for _ in range(c.num_layers)]
        ptrs = []
        for li, layer in enumerate(model.layers):
            ptrs.append([
                layer.ln1.weight.data_ptr(),
                layer.q_proj.weight.data_ptr(), layer.q_proj.bias.data_ptr(),
                layer.k_proj.weight.data_ptr(), layer.k_proj.bias.data_ptr(),
                layer.v_proj.weight.data_ptr(), layer.v_proj.bias.data_ptr(),
                layer.o_proj.weight.data_ptr(),
                layer.ln2.weight.data_ptr(),
                layer.gate_proj.weight.data_ptr(),
                layer.up_proj.weight.data_ptr(),
                layer.down_proj.weight.data_ptr(),
                self.cache_k[li].data_ptr(), self.cache_v[li].data_ptr(),
            ])
        self.layer_ptrs = torch.tensor(ptrs, dtype=torch.int64, device=dev)
        self.rope_cos = model.rope_cos.float().contiguous().to(dev)
        self.rope_sin = model.rope_sin.float().contiguous().to(dev)
        self.x = torch.zeros(c.hidden_size, device=dev, dtype=torch.float32)
        self.qs = torch.zeros_like(self.x)
        self.attn = torch.zeros_like(self.x)
        self.hbuf = torch.zeros(c.intermediate_size, device=dev,
                                dtype=torch.float32)
        # multi-token greedy path (decode_tokens): weights + scratch
        self.embed_w = model.embed.weight.data.contiguous()
        self.norm_w = model.norm.weight.data.contiguous()
        self.lm_w = model.lm_head.weight.data.contiguous()
        self.tok_out = torch.zeros(512, dtype=torch.int32, device=dev)
        self.n_done = torch.zeros(1, dtype=torch.int32, device=dev)
        self.pmax = torch.zeros(512, dtype=torch.float32, device=dev)
        self.pidx = torch.zeros(512, dtype=torch.int32, device=dev)

    @torch.no_grad()
    def step_logits(self, token_id: torch.Tensor, pos: int) -> torch.Tensor:
        """One fused decode step; returns fp32 logits [vocab]."""
        m, c = self.m, self.cfg
        self.x.copy_(m.embed.weight.index_select(
            0, token_id.view(1)).float().view(-1))
        self.nat.decode_step(self.layer_ptrs, self.x, self.qs, self.attn,
                             self.hbuf, self.rope_cos, self.rope_sin,
                             c.num_layers, c.hidden_size, c.num_heads,
                             c.num_kv_heads, self.hd, c.intermediate_size,
                             self.max_len, c.rms_eps, pos)
        xn = self.x * torch.rsqrt(self.x.pow(2).mean() + c.rms_eps)
        xn = xn * m.norm.weight.float()
        return (xn.to(torch.bfloat16) @ m.lm_head.weight.T).float()

    @torch.no_grad()
    def decode_greedy(self, start_tok: int, pos0: int, n_toks: int,
                      eos_id=None) -> list:
        """Up to n_toks greedy tokens in ONE cooperative launch.

        The kernel embeds start_tok at pos0, runs all layers, takes the
        lm_head argmax in-kernel, feeds it back, and repeats — no host
        round-trip per token. Returns the emitted token ids (shorter
        than n_toks iff EOS was emitted)."""
        c = self.cfg
        n = min(n_toks, self.max_len - pos0, self.tok_out.numel())
        if n <= 0:
            return []
        self.n_done.zero_()
        self.nat.decode_tokens(
            self.layer_ptrs, self.x, self.qs, self.attn, self.hbuf,
            self.rope_cos, self.rope_sin, self.embed_w, self.norm_w,
            self.lm_w, self.tok_out, self.n_done, self.pmax, self.pidx,
            c.num_layers, c.hidden_size, c.num_heads, c.num_kv_heads,
            self.hd, c.intermediate_size, self.max_len, c.rms_eps, pos0,
            start_tok, n, -1 if eos_id is None else int(eos_id))
        k = int(self.n_done.item())
        return self.tok_out[:k].tolist()

    @torch.no_grad()
    def generate(self, token_ids: torch.Tensor, max_new_tokens: int = 32,
                 temperature: float = 0.8, top_k: int = 40,
                 eos_id=None, stream_cb=None):
        """Prefill with the dynamic forward, then fused per-token steps."""
        m = self.m
        s = token_ids.shape[1]
        caches = [(None, None)] * len(m.layers)
        logits, caches = m.forward(token_ids, kv_caches=caches, pos0=0)
        for li, (k, v) in enumerate(caches):
            self.cache_k[li][:, :s] = k[0]
            self.cache_v[li][:, :s] = v[0]
        last = logits[:, -1, :].float().view(-1)
        out = []
        pos = s
        import os as _os
        if temperature <= 0 and _os.environ.get("NORNICDB_DECODE_INKERNEL") == "1":
            # whole greedy loop in-kernel (decode_tokens, one launch per
            # chunk). Measured SLOWER than the host loop on one stream
            # (445 vs 511 tok/s: the in-kernel lm_head runs on the
            # 64-WG cooperative grid while torch's GEMV uses the full
            # chip), but it frees the host entirely — opt-in for
            # concurrent serving. Chunked so stream_cb stays responsive.
            tok = int(last.argmax().item())
            out.append(tok)
            if stream_cb is not None:
                stream_cb(tok)
            if eos_id is not None and tok == eos_id:
                return out
            chunk = 16 if stream_cb is not None else 256
            while len(out) < max_new_tokens and pos < self.max_len:
                want = min(max_new_tokens - len(out), chunk)
                toks = self.decode_greedy(out[-1], pos, want, eos_id=eos_id)
                if not toks:
                    break
                pos += len(toks)
                for t2 in toks:
                    out.append(t2)
                    if stream_cb is not None:
                        stream_cb(t2)
                if eos_id is not None and toks[-1] == eos_id:
                    break
            return out
        cur = None
        for _ in range(max_new_tokens):
            if temperature <= 0:
                cur = last.argmax(-1, keepdim=True)
            else:
                sc = last / temperature
                if top_k:
                    vv, ix = torch.topk(sc, min(top_k, sc.shape[-1]))
                    j = torch.multinomial(torch.softmax(vv, -1), 1)
                    cur = ix.index_select(0, j.view(-1))
                else:
                    cur = torch.multinomial(torch.softmax(sc, -1), 1)
            tok = int(cur.item())
            out.append(tok)
            if stream_cb is not None:
                stream_cb(tok)
            if eos_id is not None and tok == eos_id:
                break
            if pos >= self.max_len:
                break
            last = self.step_logits(cur, pos)
            pos += 1
        return out
