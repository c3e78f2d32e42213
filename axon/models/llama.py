"""Llama-3-style decoder for the pure-DP large-model config
(BASELINE.json config 4: Llama-3 8B pure data-parallel bf16 — MI355X's
288 GB HBM holds params + grads + Adam states per GPU with no sharding,
SURVEY.md §5.7).

Own implementation of the published architecture (Grattafiori et al.,
The Llama 3 Herd of Models): RMSNorm, rotary embeddings (theta 500000),
grouped-query attention (8 KV heads), SwiGLU FFN, tied-off LM head.
"""
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from axon.ops.rmsnorm import FusedRMSNorm
from axon.ops.rotary import rope, swiglu


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    dim: int = 4096
    layers: int = 32
    heads: int = 32
    kv_heads: int = 8
    ffn: int = 14336
    max_seq: int = 8192
    rope_theta: float = 500000.0
    norm_eps: float = 1e-5


def llama3_8b_config():
    return LlamaConfig()


def llama3_1b_config():
    """Roughly Llama-3.2-1B shaped (smaller bench/dev variant)."""
    return LlamaConfig(dim=2048, layers=16, heads=32, kv_heads=8, ffn=8192)


def llama_tiny_config():
    """For tests: same code path, tiny shapes."""
    return LlamaConfig(vocab_size=256, dim=64, layers=2, heads=4, kv_heads=2,
                       ffn=128, max_seq=128)


# RMSNorm: the fused CDNA4 kernel (axon/ops/rmsnorm.py) on GPU, eager fp32
# formula elsewhere — identical math, same state dict.
RMSNorm = FusedRMSNorm


def _rope_cache(dim, seq, theta, device, dtype):
    inv = 1.0 / (theta ** (torch.arange(0, dim, 2, device=device).float() / dim))
    t = torch.arange(seq, device=device).float()
    freqs = torch.outer(t, inv)
    return torch.cos(freqs).to(dtype), torch.sin(freqs).to(dtype)


# rotary applied via the fused CDNA4 kernel (axon/ops/rotary.py) on the
# pre-transpose (b, s, h, d) layout; eager strided fallback off-GPU.


class LlamaAttention(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.heads = cfg.heads
        self.kv_heads = cfg.kv_heads
        self.head_dim = cfg.dim // cfg.heads
        self.wq = nn.Linear(cfg.dim, cfg.dim, bias=False)
        self.wk = nn.Linear(cfg.dim, cfg.kv_heads * self.head_dim, bias=False)
        self.wv = nn.Linear(cfg.dim, cfg.kv_heads * self.head_dim, bias=False)
        self.wo = nn.Linear(cfg.dim, cfg.dim, bias=False)

    def forward(self, x, cos, sin, cache=None):
        b, s, _ = x.shape
        q = rope(self.wq(x).view(b, s, self.heads, self.head_dim), cos, sin)
        k = rope(self.wk(x).view(b, s, self.kv_heads, self.head_dim), cos, sin)
        v = self.wv(x).view(b, s, self.kv_heads, self.head_dim)
        causal = True
        if cache is not None:
            # Incremental decode: append this call's K/V to the layer cache
            # ((b, t, kv_heads, d) pre-transpose layout). For the prefill
            # call the cache is empty and causal attention applies; decode
            # steps (q shorter than the cache) attend to everything past —
            # torch SDPA's is_causal aligns top-left, which would mask the
            # history, so causal turns off there.
            if cache["k"] is not None:
                k = torch.cat([cache["k"], k], 1)
                v = torch.cat([cache["v"], v], 1)
            cache["k"], cache["v"] = k, v
            causal = s == k.shape[1]
        # AXON_FA=1 routes through the CDNA4 flash-attention kernels
        # (axon/ops/flash_attention.py); default is torch SDPA.
        from axon.ops.flash_attention import flash_attention
        o = flash_attention(q.transpose(1, 2), k.transpose(1, 2),
                            v.transpose(1, 2), causal=causal)
        return self.wo(o.transpose(1, 2).reshape(b, s, -1))


class LlamaBlock(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.attn_norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.attn = LlamaAttention(cfg)
        self.ffn_norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.w1 = nn.Linear(cfg.dim, cfg.ffn, bias=False)  # gate
        self.w3 = nn.Linear(cfg.dim, cfg.ffn, bias=False)  # up
        self.w2 = nn.Linear(cfg.ffn, cfg.dim, bias=False)  # down

    def forward(self, x, cos, sin, cache=None):
        x = x + self.attn(self.attn_norm(x), cos, sin, cache)
        h = self.ffn_norm(x)
        return x + self.w2(swiglu(self.w1(h), self.w3(h)))


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.embed = nn.Embedding(cfg.vocab_size, cfg.dim)
        self.blocks = nn.ModuleList(LlamaBlock(cfg) for _ in range(cfg.layers))
        self.norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.lm_head = nn.Linear(cfg.dim, cfg.vocab_size, bias=False)
        self.apply(self._init)
        self._rope = {}

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)

    def _rope_for(self, s, device, dtype):
        key = (device, dtype)
        have = self._rope.get(key)
        if have is None or have[0].shape[0] < s:
            # Grow geometrically so incremental decoding (s+1 per step)
            # doesn't rebuild the table every token.
            n = max(s, 2 * have[0].shape[0] if have is not None else s)
            self._rope[key] = _rope_cache(self.cfg.dim // self.cfg.heads, n,
                                          self.cfg.rope_theta, device, dtype)
        cos, sin = self._rope[key]
        return cos[:s], sin[:s]

    def forward(self, input_ids, caches=None, pos=0):
        b, s = input_ids.shape
        x = self.embed(input_ids)
        # Rope rows for the ABSOLUTE positions of these tokens (pos offset
        # when decoding against a KV cache).
        cos, sin = self._rope_for(pos + s, x.device, x.dtype)
        cos, sin = cos[pos:pos + s], sin[pos:pos + s]
        for i, blk in enumerate(self.blocks):
            x = blk(x, cos, sin, caches[i] if caches is not None else None)
        return self.lm_head(self.norm(x))

    def lm_loss(self, input_ids):
        """Next-token loss on the sequence itself (synthetic pretrain)."""
        logits = self.forward(input_ids)
        return F.cross_entropy(
            logits[:, :-1].float().reshape(-1, self.cfg.vocab_size),
            input_ids[:, 1:].reshape(-1))

    @torch.no_grad()
    def generate(self, input_ids, max_new_tokens, temperature=0.0,
                 top_k=None):
        """KV-cached autoregressive decoding: one prefill pass over the
        prompt, then one single-token pass per generated token (each decode
        step reads the per-layer K/V cache instead of recomputing the
        prefix). temperature 0 = greedy; otherwise softmax sampling over
        the top_k (or full) distribution."""
        caches = [{"k": None, "v": None} for _ in self.blocks]
        tokens = input_ids
        step_in = input_ids
        pos = 0
        for _ in range(max_new_tokens):
            logits = self.forward(step_in, caches, pos)[:, -1].float()
            pos += step_in.shape[1]
            if temperature > 0:
                logits = logits / temperature
                if top_k is not None:
                    kth = torch.topk(logits, top_k, dim=-1).values[..., -1:]
                    logits = logits.masked_fill(logits < kth,
                                                float("-inf"))
                nxt = torch.multinomial(F.softmax(logits, -1), 1)
            else:
                nxt = logits.argmax(-1, keepdim=True)
            tokens = torch.cat([tokens, nxt], 1)
            step_in = nxt
        return tokens
