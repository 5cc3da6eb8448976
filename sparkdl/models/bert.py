"""BERT-base for the seq-512 bf16 pretrain benchmark (BASELINE.json
config 4), written from scratch.

MI355X mapping (round 2): every Linear — QKV, attention out, both FFN
halves, the MLM dense — runs on the hand-written 256x256 16-wave MFMA
GEMM with fused bias(+GELU) epilogues and in-house dgrad; attention is
the hand-written flash forward+backward (packed-qkv zero-copy I/O,
in-kernel dropout) — no Triton/aotriton and no library GEMM on the
fwd/dgrad path (wgrad remains hipBLASLt, see NOTES-round2.md);
LayerNorm and the optimizer (fused multi-tensor AdamW) are hand-written
CDNA4 kernels.
"""

import math

import torch
import torch.nn as nn

from sparkdl.ops import LayerNorm, Linear, LinearGelu


class BertConfig:
    def __init__(self, vocab_size=30522, hidden=768, layers=12, heads=12,
                 ffn=3072, max_seq=512, type_vocab=2, dropout=0.1):
        self.vocab_size = vocab_size
        self.hidden = hidden
        self.layers = layers
        self.heads = heads
        self.ffn = ffn
        self.max_seq = max_seq
        self.type_vocab = type_vocab
        self.dropout = dropout


class BertEmbeddings(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.word = nn.Embedding(cfg.vocab_size, cfg.hidden)
        self.position = nn.Embedding(cfg.max_seq, cfg.hidden)
        self.token_type = nn.Embedding(cfg.type_vocab, cfg.hidden)
        self.ln = LayerNorm(cfg.hidden, eps=1e-12)
        self.drop = nn.Dropout(cfg.dropout)

    def forward(self, ids, token_type=None):
        B, S = ids.shape
        pos = torch.arange(S, device=ids.device).unsqueeze(0)
        h = self.word(ids) + self.position(pos)
        if token_type is not None:
            h = h + self.token_type(token_type)
        return self.drop(self.ln(h))


class BertSelfAttention(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.heads = cfg.heads
        self.head_dim = cfg.hidden // cfg.heads
        self.qkv = Linear(cfg.hidden, 3 * cfg.hidden)
        self.out = Linear(cfg.hidden, cfg.hidden)
        self.drop_p = cfg.dropout

    def forward(self, x):
        from sparkdl.ops import functional as F_
        B, S, H = x.shape
        qkv = self.qkv(x).view(B, S, 3, self.heads, self.head_dim)
        p = self.drop_p if self.training else 0.0
        if (qkv.is_cuda and qkv.dtype == torch.bfloat16
                and self.head_dim == 64 and S % 64 == 0):
            # hand-written CDNA4 flash kernels (fwd+bwd, in-kernel
            # dropout), reading the packed qkv buffer with strided rows
            # and writing O as [B,S,H] directly — no Triton and no
            # permute/contiguous copies on the hot path
            o = F_.flash_attention_packed(qkv, dropout_p=p)
        else:
            q, k, v = qkv.permute(2, 0, 3, 1, 4).unbind(0)  # [B,h,S,d]
            o = nn.functional.scaled_dot_product_attention(
                q, k, v, dropout_p=p)
            o = o.transpose(1, 2).reshape(B, S, H)
        return self.out(o)


class BertLayer(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.attn = BertSelfAttention(cfg)
        self.ln1 = LayerNorm(cfg.hidden, eps=1e-12)
        self.ffn_in = LinearGelu(cfg.hidden, cfg.ffn)
        self.ffn_out = Linear(cfg.ffn, cfg.hidden)
        self.ln2 = LayerNorm(cfg.hidden, eps=1e-12)
        self.drop = nn.Dropout(cfg.dropout)

    def forward(self, x):
        x = self.ln1(x + self.drop(self.attn(x)))
        x = self.ln2(x + self.drop(self.ffn_out(self.ffn_in(x))))
        return x


class BertBase(nn.Module):
    def __init__(self, cfg=None):
        super().__init__()
        self.cfg = cfg = cfg or BertConfig()
        self.embeddings = BertEmbeddings(cfg)
        self.encoder = nn.ModuleList(
            [BertLayer(cfg) for _ in range(cfg.layers)])
        # MLM head with tied decoder weights.
        self.mlm_dense = Linear(cfg.hidden, cfg.hidden)
        self.mlm_ln = LayerNorm(cfg.hidden, eps=1e-12)
        self.mlm_bias = nn.Parameter(torch.zeros(cfg.vocab_size))
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, Linear, LinearGelu)):
            nn.init.normal_(m.weight, std=0.02)
            if getattr(m, "bias", None) is not None:
                nn.init.zeros_(m.bias)
        elif isinstance(m, nn.Embedding):
            nn.init.normal_(m.weight, std=0.02)

    def encode(self, ids, token_type=None):
        h = self.embeddings(ids, token_type)
        # Keep the residual stream in the autocast compute dtype (bf16) so
        # every LayerNorm / bias+GELU hits the CDNA4 kernels; embeddings
        # and type-promoted residual adds would otherwise pin it to fp32.
        if h.is_cuda and torch.is_autocast_enabled():
            h = h.to(torch.get_autocast_dtype("cuda"))
        for layer in self.encoder:
            h = layer(h)
        return h

    def _mlm_logits(self, h):
        h = self.mlm_ln(torch.nn.functional.gelu(self.mlm_dense(h)))
        logits = torch.nn.functional.linear(
            h, self.embeddings.word.weight.to(h.dtype), None)
        return logits + self.mlm_bias.to(logits.dtype)

    def forward(self, ids, token_type=None):
        return self._mlm_logits(self.encode(ids, token_type))

    def forward_mlm(self, ids, positions, token_type=None):
        """MLM logits only at the masked positions (flat indices into the
        [B*S] token stream) — the vocab projection runs on ~15% of tokens
        instead of all of them, the standard pretrain-head optimization."""
        h = self.encode(ids, token_type)
        h = h.reshape(-1, self.cfg.hidden).index_select(0, positions)
        return self._mlm_logits(h)


def bert_pretrain_step(model, opt, batch, seq, device, use_cuda,
                       mask_frac=0.15):
    """Build a closure running one synthetic MLM pretrain step.

    The vocab projection + loss run only on the masked positions
    (mask_frac of tokens), as in real MLM pretraining.
    """
    cfg = model.cfg
    g = torch.Generator(device="cpu").manual_seed(7)
    ids = torch.randint(0, cfg.vocab_size, (batch, seq), generator=g) \
        .to(device)
    mask = torch.rand(batch, seq, generator=g) < mask_frac
    positions = mask.flatten().nonzero(as_tuple=False).flatten().to(device)
    labels = ids.flatten().index_select(0, positions)
    autocast_dev = "cuda" if use_cuda else "cpu"
    # Pure-bf16 weights (convert_bf16_training) need no autocast: every
    # GEMM already sees bf16 operands and grads flow in bf16.
    pure_bf16 = model.embeddings.word.weight.dtype == torch.bfloat16

    def step():
        opt.zero_grad()
        if pure_bf16:
            logits = model.forward_mlm(ids, positions)
            loss = torch.nn.functional.cross_entropy(
                logits.float(), labels)
        else:
            with torch.autocast(autocast_dev, dtype=torch.bfloat16):
                logits = model.forward_mlm(ids, positions)
                loss = torch.nn.functional.cross_entropy(
                    logits.float(), labels)
        loss.backward()
        opt.step()
        return loss

    return step
