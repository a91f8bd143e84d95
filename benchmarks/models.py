# -*- coding: utf-8 -*-
"""Benchmark model zoo (self-contained; no torchvision/transformers weights).

Implements the architectures named by BASELINE.json:
* ResNet-18/50 (He et al. 2015 bottleneck design) for the CPU plumbing and
  DDP-bf16 headline benches,
* GPT-2-medium-shape decoder for the OSS (ZeRO-1) bench,
* Llama-3-8B-shape decoder (RMSNorm + RoPE + SwiGLU + GQA) for the FSDP bench.

All models random-init; benches use synthetic data (no network access).
"""

import math
from typing import List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F


# ---------------------------------------------------------------------------
# ResNet
# ---------------------------------------------------------------------------
from stoke.nn import FusedBNAct2d


class _Downsample(nn.Module):
    def __init__(self, cin, cout, stride):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, 1, stride, bias=False)
        self.bn = FusedBNAct2d(cout, relu=False)

    def forward(self, x):
        return self.bn(self.conv(x))


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin, cout, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(cin, cout, 3, stride, 1, bias=False)
        self.bn1 = FusedBNAct2d(cout, relu=True)
        self.conv2 = nn.Conv2d(cout, cout, 3, 1, 1, bias=False)
        # bn2 fuses the residual add and final ReLU into one kernel pass
        self.bn2 = FusedBNAct2d(cout, relu=True)
        self.downsample = downsample

    def forward(self, x):
        idt = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        return self.bn2(self.conv2(out), residual=idt)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, cmid, stride=1, downsample=None):
        super().__init__()
        cout = cmid * self.expansion
        self.conv1 = nn.Conv2d(cin, cmid, 1, 1, 0, bias=False)
        self.bn1 = FusedBNAct2d(cmid, relu=True)
        self.conv2 = nn.Conv2d(cmid, cmid, 3, stride, 1, bias=False)
        self.bn2 = FusedBNAct2d(cmid, relu=True)
        self.conv3 = nn.Conv2d(cmid, cout, 1, 1, 0, bias=False)
        self.bn3 = FusedBNAct2d(cout, relu=True)
        self.downsample = downsample

    def forward(self, x):
        idt = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        return self.bn3(self.conv3(out), residual=idt)


class ResNet(nn.Module):
    def __init__(self, block, layers: List[int], num_classes=1000, small_input=False):
        super().__init__()
        self.cin = 64
        if small_input:  # CIFAR-shape stem
            self.stem = nn.Sequential(
                nn.Conv2d(3, 64, 3, 1, 1, bias=False),
                FusedBNAct2d(64, relu=True),
            )
        else:
            self.stem = nn.Sequential(
                nn.Conv2d(3, 64, 7, 2, 3, bias=False),
                FusedBNAct2d(64, relu=True),
                nn.MaxPool2d(3, 2, 1),
            )
        self.layer1 = self._make_layer(block, 64, layers[0], 1)
        self.layer2 = self._make_layer(block, 128, layers[1], 2)
        self.layer3 = self._make_layer(block, 256, layers[2], 2)
        self.layer4 = self._make_layer(block, 512, layers[3], 2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * block.expansion, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
            elif isinstance(m, (nn.BatchNorm2d, FusedBNAct2d)):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def _make_layer(self, block, cmid, n, stride):
        downsample = None
        cout = cmid * block.expansion
        if stride != 1 or self.cin != cout:
            downsample = _Downsample(self.cin, cout, stride)
        layers = [block(self.cin, cmid, stride, downsample)]
        self.cin = cout
        for _ in range(1, n):
            layers.append(block(self.cin, cmid))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.stem(x)
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet18(num_classes=1000, small_input=False):
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes, small_input)


def resnet50(num_classes=1000, small_input=False):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, small_input)


def resnet152(num_classes=1000, small_input=False):
    return ResNet(Bottleneck, [3, 8, 36, 3], num_classes, small_input)


# ---------------------------------------------------------------------------
# GPT-2-shape decoder (learned positions, LayerNorm, GELU MLP)
# ---------------------------------------------------------------------------
class GPT2Block(nn.Module):
    """Pre-LN GPT-2 block with fused QKV + causal flash SDPA.

    (An earlier version used nn.MultiheadAttention with an additive mask,
    which falls off the SDPA fast path and materializes S x S weights.)
    """

    def __init__(self, d, nh, dropout=0.0):
        super().__init__()
        from stoke.nn import StokeLayerNorm

        self.nh = nh
        self.ln1 = StokeLayerNorm(d)
        self.qkv = nn.Linear(d, 3 * d)
        self.proj = nn.Linear(d, d)
        self.ln2 = StokeLayerNorm(d)
        self.mlp = nn.Sequential(
            nn.Linear(d, 4 * d), nn.GELU(), nn.Linear(4 * d, d)
        )

    def forward(self, x):
        B, S, D = x.shape
        h = self.ln1(x)
        q, k, v = self.qkv(h).split(D, dim=-1)
        q = q.view(B, S, self.nh, -1).transpose(1, 2)
        k = k.view(B, S, self.nh, -1).transpose(1, 2)
        v = v.view(B, S, self.nh, -1).transpose(1, 2)
        from stoke.nn.attention import attention

        a = attention(q, k, v, causal=True)
        x = x + self.proj(a.transpose(1, 2).reshape(B, S, D))
        return x + self.mlp(self.ln2(x))


class GPT2(nn.Module):
    """GPT-2 family; defaults are the "medium" shape (355M params)."""

    def __init__(self, vocab=50257, d=1024, nlayer=24, nh=16, max_seq=1024):
        super().__init__()
        self.wte = nn.Embedding(vocab, d)
        self.wpe = nn.Embedding(max_seq, d)
        from stoke.nn import StokeLayerNorm

        self.blocks = nn.ModuleList([GPT2Block(d, nh) for _ in range(nlayer)])
        self.ln_f = StokeLayerNorm(d)
        self.head = nn.Linear(d, vocab, bias=False)
        self.head.weight = self.wte.weight  # tied
        self.max_seq = max_seq

    def forward(self, idx):
        B, S = idx.shape
        pos = torch.arange(S, device=idx.device)
        x = self.wte(idx) + self.wpe(pos)[None]
        for blk in self.blocks:
            x = blk(x)
        return self.head(self.ln_f(x))


def gpt2_medium(vocab=50257, max_seq=1024):
    return GPT2(vocab=vocab, d=1024, nlayer=24, nh=16, max_seq=max_seq)


# ---------------------------------------------------------------------------
# Llama-3-shape decoder (RMSNorm, RoPE, SwiGLU, GQA) — sized for 8B default
# ---------------------------------------------------------------------------
# HIP-fused on bf16 GPU inputs, eager fp32 composition elsewhere
from stoke.nn import StokeRMSNorm as RMSNorm  # noqa: E402


def _rope_cache(seq, hd, device, base=500000.0):
    inv = 1.0 / (base ** (torch.arange(0, hd, 2, device=device).float() / hd))
    t = torch.arange(seq, device=device).float()
    freqs = torch.outer(t, inv)
    return freqs.cos(), freqs.sin()


# HIP-fused on bf16 GPU inputs ([B, S, H, Dh] layout), eager elsewhere
from stoke.nn.rope import apply_rope  # noqa: E402


class LlamaBlock(nn.Module):
    def __init__(self, d, nh, nkv, ffn):
        super().__init__()
        self.nh, self.nkv = nh, nkv
        self.hd = d // nh
        self.attn_norm = RMSNorm(d)
        self.wq = nn.Linear(d, nh * self.hd, bias=False)
        self.wk = nn.Linear(d, nkv * self.hd, bias=False)
        self.wv = nn.Linear(d, nkv * self.hd, bias=False)
        self.wo = nn.Linear(nh * self.hd, d, bias=False)
        self.mlp_norm = RMSNorm(d)
        self.w_gate = nn.Linear(d, ffn, bias=False)
        self.w_up = nn.Linear(d, ffn, bias=False)
        self.w_down = nn.Linear(ffn, d, bias=False)

    def forward(self, x, cos, sin):
        B, S, D = x.shape
        h = self.attn_norm(x)
        # RoPE on the contiguous [B, S, H, Dh] projections (one fused kernel
        # each), then the attention transpose
        q = apply_rope(self.wq(h).view(B, S, self.nh, self.hd), cos, sin)
        k = apply_rope(self.wk(h).view(B, S, self.nkv, self.hd), cos, sin)
        q = q.transpose(1, 2)
        k = k.transpose(1, 2)
        v = self.wv(h).view(B, S, self.nkv, self.hd).transpose(1, 2)
        from stoke.nn.attention import attention

        a = attention(q, k, v, causal=True)
        a = a.transpose(1, 2).reshape(B, S, -1)
        x = x + self.wo(a)
        h = self.mlp_norm(x)
        from stoke.nn import swiglu

        return x + self.w_down(swiglu(self.w_gate(h), self.w_up(h)))


class Llama(nn.Module):
    """Llama-3 family; defaults are the 8B shape."""

    def __init__(
        self,
        vocab=128256,
        d=4096,
        nlayer=32,
        nh=32,
        nkv=8,
        ffn=14336,
        max_seq=8192,
    ):
        super().__init__()
        self.embed = nn.Embedding(vocab, d)
        self.blocks = nn.ModuleList(
            [LlamaBlock(d, nh, nkv, ffn) for _ in range(nlayer)]
        )
        self.norm = RMSNorm(d)
        self.head = nn.Linear(d, vocab, bias=False)
        self.hd = d // nh
        self.max_seq = max_seq
        self._cos = None
        self._sin = None

    def forward(self, idx):
        B, S = idx.shape
        if self._cos is None or self._cos.shape[0] < S or self._cos.device != idx.device:
            self._cos, self._sin = _rope_cache(max(S, 2048), self.hd, idx.device)
        x = self.embed(idx)
        for blk in self.blocks:
            x = blk(x, self._cos, self._sin)
        return self.head(self.norm(x))


def llama3_8b(vocab=128256, max_seq=8192):
    return Llama(vocab=vocab, d=4096, nlayer=32, nh=32, nkv=8, ffn=14336,
                 max_seq=max_seq)


def llama_tiny(vocab=1024, d=256, nlayer=2, nh=4, nkv=2, ffn=512):
    """Small shape for CPU tests."""
    return Llama(vocab=vocab, d=d, nlayer=nlayer, nh=nh, nkv=nkv, ffn=ffn,
                 max_seq=512)
