"""MI355X-native T5 / FLAN-T5 encoder-decoder.

Re-implements the model family the reference workshop fine-tunes and serves
(``T5ForConditionalGeneration`` — reference call sites
Model_finetuning_and_batch_inference.ipynb:cell-25/cell-46/cell-49,
NLP_workloads/Anyscale_job/predictor.py:102) from scratch on top of
``mi355.ops``: every hot op (RMSNorm, fused attention with T5's additive
relative-position bias, gated-GELU, vocab cross-entropy) dispatches to
hand-written HIP/CDNA4 kernels on GPU; plain GEMMs go through torch's
hipBLASLt path. Parameter names and the ``save_pretrained`` checkpoint layout
match the HF T5 convention so checkpoints are drop-in.
"""
from __future__ import annotations

import math
import os
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from .configuration import T5Config, t5_config

NEG_INF = -1e9


class T5LayerNorm(nn.Module):
    def __init__(self, d: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(d))
        self.variance_epsilon = eps

    def forward(self, x):
        return ops.rms_norm(x, self.weight, self.variance_epsilon)


def relative_position_bucket(rel_pos: torch.Tensor, bidirectional: bool,
                             num_buckets: int, max_distance: int) -> torch.Tensor:
    """T5 bucketing of (key_pos - query_pos) — SURVEY.md K4."""
    ret = torch.zeros_like(rel_pos)
    n = -rel_pos
    if bidirectional:
        num_buckets //= 2
        ret = ret + (n < 0).long() * num_buckets
        n = n.abs()
    else:
        n = torch.clamp(n, min=0)
    max_exact = num_buckets // 2
    is_small = n < max_exact
    val_large = max_exact + (
        torch.log(n.float() / max_exact + 1e-20) / math.log(max_distance / max_exact)
        * (num_buckets - max_exact)
    ).long()
    val_large = torch.clamp(val_large, max=num_buckets - 1)
    return ret + torch.where(is_small, n, val_large)


class _FusedLinears(nn.Module):
    """N same-shaped bias-free projections stored as ONE fused weight so the
    forward is a single hipBLASLt GEMM (q/k/v -> 1 GEMM, wi_0/wi_1 -> 1 GEMM:
    the small-M projection GEMMs at B=2 are launch/M-bound).
    state_dict stays HF-compatible: each sub-projection saves/loads under its
    own key as a slice view."""

    def __init__(self, names, d_in, d_out):
        super().__init__()
        self.names = list(names)
        self.d_out = d_out
        self.weight = nn.Parameter(torch.empty(len(self.names) * d_out, d_in))

    def forward(self, x):
        y = F.linear(x, self.weight)
        return y.split(self.d_out, dim=-1)

    def part(self, i):
        return self.weight[i * self.d_out:(i + 1) * self.d_out]


def _register_fused_statedict_compat(parent: nn.Module, attr: str):
    """Keep HF key names for a _FusedLinears child: on save, split
    ``<prefix><attr>.weight`` into ``<prefix><name>.weight`` slices; on load,
    merge HF per-projection keys back into the fused weight."""
    fused: _FusedLinears = getattr(parent, attr)
    names = list(fused.names)

    def save_hook(module, state_dict, prefix, local_metadata):
        key = f"{prefix}{attr}.weight"
        if key in state_dict:
            w = state_dict.pop(key)
            for i, n in enumerate(names):
                state_dict[f"{prefix}{n}.weight"] = \
                    w[i * fused.d_out:(i + 1) * fused.d_out]
        return state_dict

    def load_pre_hook(state_dict, prefix, local_metadata, strict, missing_keys,
                      unexpected_keys, error_msgs):
        part_keys = [f"{prefix}{n}.weight" for n in names]
        if all(k in state_dict for k in part_keys):
            state_dict[f"{prefix}{attr}.weight"] = torch.cat(
                [state_dict.pop(k) for k in part_keys], dim=0)

    parent._register_state_dict_hook(save_hook)
    parent._register_load_state_dict_pre_hook(load_pre_hook)


def _register_cross_kv_compat(stack: nn.Module, n_layers: int, inner: int):
    """HF key compatibility for the decoder-level fused cross-kv parameter:
    on save, split ``cross_kv`` into the per-layer
    ``block.{i}.layer.1.EncDecAttention.{k,v}.weight`` slices; on load, merge
    those HF keys back into the one fused weight."""

    def save_hook(module, state_dict, prefix, local_metadata):
        key = f"{prefix}cross_kv"
        if key in state_dict:
            w = state_dict.pop(key)
            for i in range(n_layers):
                blk = w[i * 2 * inner:(i + 1) * 2 * inner]
                base = f"{prefix}block.{i}.layer.1.EncDecAttention"
                state_dict[f"{base}.k.weight"] = blk[:inner]
                state_dict[f"{base}.v.weight"] = blk[inner:]
        return state_dict

    def load_pre_hook(state_dict, prefix, local_metadata, strict, missing_keys,
                      unexpected_keys, error_msgs):
        pairs = []
        for i in range(n_layers):
            base = f"{prefix}block.{i}.layer.1.EncDecAttention"
            pairs.append((f"{base}.k.weight", f"{base}.v.weight"))
        if all(k in state_dict and v in state_dict for k, v in pairs):
            state_dict[f"{prefix}cross_kv"] = torch.cat(
                [torch.cat([state_dict.pop(k), state_dict.pop(v)], dim=0)
                 for k, v in pairs], dim=0)

    stack._register_state_dict_hook(save_hook)
    stack._register_load_state_dict_pre_hook(load_pre_hook)


class T5Attention(nn.Module):
    def __init__(self, config: T5Config, has_relative_attention_bias: bool, is_decoder: bool,
                 kv_external: bool = False):
        super().__init__()
        self.is_decoder = is_decoder
        self.has_relative_attention_bias = has_relative_attention_bias
        self.num_buckets = config.relative_attention_num_buckets
        self.max_distance = config.relative_attention_max_distance
        self.d_model = config.d_model
        self.d_kv = config.d_kv
        self.n_heads = config.num_heads
        self.inner_dim = self.n_heads * self.d_kv
        # kv_external (decoder cross-attention): the k/v weights of ALL
        # layers live in ONE decoder-level parameter (T5Stack.cross_kv) so
        # the 12 per-layer kv projections over the SAME encoder output become
        # a single (B*S, 12*2*inner) GEMM per forward; this module keeps only
        # its q projection (HF state-dict keys are preserved by the stack's
        # save/load hooks).
        self.kv_external = kv_external
        names = ["q"] if kv_external else ["q", "k", "v"]
        self.qkv = _FusedLinears(names, self.d_model, self.inner_dim)
        _register_fused_statedict_compat(self, "qkv")
        self.o = nn.Linear(self.inner_dim, self.d_model, bias=False)
        if has_relative_attention_bias:
            self.relative_attention_bias = nn.Embedding(self.num_buckets, self.n_heads)
        self.dropout = config.dropout_rate

    # q/k/v as single projections (cross-attention + generate cache paths)
    def q(self, x):
        return F.linear(x, self.qkv.part(0))

    def k(self, x):
        return F.linear(x, self.qkv.part(1))

    def v(self, x):
        return F.linear(x, self.qkv.part(2))

    def compute_bias(self, sq: int, sk: int, device) -> torch.Tensor:
        """(1, H, sq, sk) additive relative-position bias."""
        buckets = self._buckets(sq, sk, device)
        return ops.rel_position_bias(buckets, self.relative_attention_bias.weight)

    def _buckets(self, sq: int, sk: int, device) -> torch.Tensor:
        # bucket table depends only on shape — cache it across steps
        key = (sq, sk, str(device))
        cached = getattr(self, "_bucket_cache", None)
        if cached is not None and cached[0] == key:
            return cached[1]
        cpos = torch.arange(sq, dtype=torch.long, device=device)[:, None]
        mpos = torch.arange(sk, dtype=torch.long, device=device)[None, :]
        buckets = relative_position_bucket(
            mpos - cpos, bidirectional=not self.is_decoder,
            num_buckets=self.num_buckets, max_distance=self.max_distance,
        ).to(torch.int32).contiguous()
        self._bucket_cache = (key, buckets)
        return buckets

    def _shape(self, x, b):
        return x.view(b, -1, self.n_heads, self.d_kv).transpose(1, 2)

    def forward(self, hidden, kv_hidden=None, position_bias=None, kpad=None,
                causal=False, kv_cache=None, kv5=None):
        """kv_cache: optional dict used by generate() —
        {"k","v": (B,H,Tmax,D) preallocated, "len": int} for self-attn append,
        or {"k","v" precomputed static} for cross-attn.
        kv5: (B, Sk, 2, H, D) precomputed packed kv (decoder cross-attention
        with kv_external: one stack-level GEMM feeds every layer)."""
        b, sq, _ = hidden.shape
        static = kv_cache is not None and kv_cache.get("static", False)
        if kv_hidden is None and kv_cache is None:
            # self-attention without a cache (training / encoder): ONE fused
            # qkv GEMM consumed PACKED — the attention grad comes back as one
            # (B,S,3,H,D) tensor, so autograd never cats dq/dk/dv
            qkv5 = ops.linear(hidden, self.qkv.weight).view(
                b, sq, 3, self.n_heads, self.d_kv)
            p_drop = self.dropout if self.training else 0.0
            out = ops.attention_qkv_packed(qkv5, bias=position_bias,
                                           causal=causal, kpad=kpad,
                                           p_drop=p_drop)
            out = out.transpose(1, 2).reshape(b, sq, self.inner_dim)
            return ops.linear(out, self.o.weight)
        if kv_hidden is None and not static:
            # self-attention with a decode cache: fused qkv GEMM, split views
            q3, k3, v3 = self.qkv(hidden)
            qs = self._shape(q3, b)
            ks = self._shape(k3, b)
            vs = self._shape(v3, b)
            if kv_cache is not None:
                t0 = kv_cache["len"]
                kv_cache["k"][:, :, t0:t0 + sq] = ks
                kv_cache["v"][:, :, t0:t0 + sq] = vs
                kv_cache["len"] = t0 + sq
                ks = kv_cache["k"][:, :, : t0 + sq]
                vs = kv_cache["v"][:, :, : t0 + sq]
        elif static:
            qs = self._shape(self.q(hidden), b)
            ks, vs = kv_cache["k"], kv_cache["v"]
        else:
            # cross-attention (training): packed kv — either precomputed at
            # the stack level (kv_external, one GEMM for all 12 layers) or a
            # per-layer fused kv GEMM over the last two thirds of qkv
            qs = self._shape(self.q(hidden), b)
            if kv5 is None:
                assert not self.kv_external, \
                    "kv_external cross-attention needs the stack-level kv5"
                kv5 = ops.linear(kv_hidden, self.qkv.weight[self.inner_dim:]).view(
                    b, -1, 2, self.n_heads, self.d_kv)
            p_drop = self.dropout if self.training else 0.0
            out = ops.attention_kv_packed(qs, kv5, bias=position_bias,
                                          kpad=kpad, p_drop=p_drop)
            out = out.transpose(1, 2).reshape(b, sq, self.inner_dim)
            return ops.linear(out, self.o.weight)
        p_drop = self.dropout if self.training else 0.0
        if kv_cache is not None and sq == 1 and not static:
            out = ops.decode_attention(qs, kv_cache["k"], kv_cache["v"],
                                       kv_cache["len"], bias=position_bias)
        elif static and sq == 1 and not torch.is_grad_enabled():
            # single-token cross-attention over the full encoder KV
            out = ops.decode_attention(qs, ks, vs, ks.shape[2],
                                       bias=position_bias, kpad=kpad)
        else:
            out = ops.attention(qs, ks, vs, bias=position_bias, causal=causal,
                                kpad=kpad, p_drop=p_drop)
        out = out.transpose(1, 2).reshape(b, sq, self.inner_dim)
        return ops.linear(out, self.o.weight)


class T5LayerSelfAttention(nn.Module):
    def __init__(self, config, has_relative_attention_bias, is_decoder):
        super().__init__()
        self.SelfAttention = T5Attention(config, has_relative_attention_bias, is_decoder)
        self.layer_norm = T5LayerNorm(config.d_model, config.layer_norm_epsilon)
        self.dropout = nn.Dropout(config.dropout_rate)

    def forward(self, hidden, position_bias=None, kpad=None, causal=False, kv_cache=None):
        normed = self.layer_norm(hidden)
        att = self.SelfAttention(normed, position_bias=position_bias, kpad=kpad,
                                 causal=causal, kv_cache=kv_cache)
        return hidden + self.dropout(att)


class T5LayerCrossAttention(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.EncDecAttention = T5Attention(config, False, is_decoder=True,
                                           kv_external=True)
        self.layer_norm = T5LayerNorm(config.d_model, config.layer_norm_epsilon)
        self.dropout = nn.Dropout(config.dropout_rate)

    def forward(self, hidden, enc_hidden, kpad=None, kv_cache=None, kv5=None):
        normed = self.layer_norm(hidden)
        att = self.EncDecAttention(normed, kv_hidden=enc_hidden, kpad=kpad,
                                   kv_cache=kv_cache, kv5=kv5)
        return hidden + self.dropout(att)


class T5DenseGatedActDense(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.wi = _FusedLinears(["wi_0", "wi_1"], config.d_model, config.d_ff)
        _register_fused_statedict_compat(self, "wi")
        self.wo = nn.Linear(config.d_ff, config.d_model, bias=False)
        self.dropout = nn.Dropout(config.dropout_rate)

    def forward(self, x):
        # packed: one fused-wi GEMM -> one GeGLU kernel with the FF dropout
        # fused on its output (no gate/up split views, no grad cat)
        p = self.dropout.p if self.training else 0.0
        h = ops.geglu_packed(ops.linear(x, self.wi.weight), p_drop=p,
                             salt=self.wi.weight.data_ptr())
        return ops.linear(h, self.wo.weight)


class T5LayerFF(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.DenseReluDense = T5DenseGatedActDense(config)
        self.layer_norm = T5LayerNorm(config.d_model, config.layer_norm_epsilon)
        self.dropout = nn.Dropout(config.dropout_rate)

    def forward(self, hidden):
        return hidden + self.dropout(self.DenseReluDense(self.layer_norm(hidden)))


class T5Block(nn.Module):
    def __init__(self, config, has_relative_attention_bias, is_decoder):
        super().__init__()
        self.is_decoder = is_decoder
        self.layer = nn.ModuleList()
        self.layer.append(T5LayerSelfAttention(config, has_relative_attention_bias, is_decoder))
        if is_decoder:
            self.layer.append(T5LayerCrossAttention(config))
        self.layer.append(T5LayerFF(config))

    def forward(self, hidden, position_bias=None, kpad=None, enc_hidden=None,
                enc_kpad=None, causal=False, self_cache=None, cross_cache=None):
        hidden = self.layer[0](hidden, position_bias=position_bias, kpad=kpad,
                               causal=causal, kv_cache=self_cache)
        if self.is_decoder:
            hidden = self.layer[1](hidden, enc_hidden, kpad=enc_kpad, kv_cache=cross_cache)
        hidden = self.layer[-1](hidden)
        return hidden


class T5Stack(nn.Module):
    def __init__(self, config: T5Config, embed_tokens: nn.Embedding, is_decoder: bool):
        super().__init__()
        self.config = config
        self.is_decoder = is_decoder
        self.embed_tokens = embed_tokens
        n = config.num_decoder_layers if is_decoder else config.num_layers
        self.block = nn.ModuleList(
            [T5Block(config, has_relative_attention_bias=(i == 0), is_decoder=is_decoder)
             for i in range(n)]
        )
        self.final_layer_norm = T5LayerNorm(config.d_model, config.layer_norm_epsilon)
        self.dropout = nn.Dropout(config.dropout_rate)
        if is_decoder:
            # Cross-layer fused kv projection: every decoder layer's
            # cross-attention projects the SAME encoder output, so the 12
            # per-layer (M, 2*inner) GEMMs collapse into one
            # (M, n*2*inner) GEMM per forward (round-1 ROADMAP item;
            # SURVEY K5).  Layout: rows [layer][k|v][inner]; the save/load
            # hooks below keep the HF per-layer k/v key names.
            inner = config.num_heads * config.d_kv
            self.cross_kv = nn.Parameter(torch.empty(n * 2 * inner, config.d_model))
            _register_cross_kv_compat(self, n, inner)

    def forward(self, input_ids=None, inputs_embeds=None, kpad=None, enc_hidden=None,
                enc_kpad=None, caches=None, past_len: int = 0):
        if inputs_embeds is None:
            inputs_embeds = ops.token_embedding(input_ids, self.embed_tokens.weight)
        x = self.dropout(inputs_embeds)
        sq = x.shape[1]
        sk = past_len + sq
        # The relative-position bias is computed once by block 0 and shared by
        # every layer (T5 semantics).
        attn0 = self.block[0].layer[0].SelfAttention
        position_bias = attn0.compute_bias(sk, sk, x.device)[:, :, past_len:].to(x.dtype) \
            if attn0.has_relative_attention_bias else None
        causal = self.is_decoder and sq > 1
        # pre-norm chain with fused residual-add + RMSNorm (SURVEY.md K2+K8):
        # resid/delta threading — each sublayer's input norm absorbs the
        # previous sublayer's residual add in one kernel
        resid, delta = x, None
        pdrop = self.config.dropout_rate if self.training else 0.0

        def add_norm(ln, resid, delta):
            if delta is None:
                return resid, ln(resid)
            # the sublayer-output dropout is fused into the add+norm kernel
            return ops.add_rms_norm(resid, delta, ln.weight, ln.variance_epsilon,
                                    p_drop=pdrop)

        # one fused kv projection for every layer's cross-attention (the
        # unbind keeps the backward a single stack instead of 12 per-layer
        # zero+scatter slice grads)
        kv_layers = None
        if self.is_decoder and enc_hidden is not None and caches is None:
            h, dkv = self.config.num_heads, self.config.d_kv
            be, se, _ = enc_hidden.shape
            kv_all = ops.linear(enc_hidden, self.cross_kv).view(
                be, se, len(self.block), 2, h, dkv)
            kv_layers = kv_all.unbind(2)

        for i, blk in enumerate(self.block):
            self_cache = caches[i]["self"] if caches is not None else None
            cross_cache = caches[i]["cross"] if (caches is not None and self.is_decoder) else None
            sa = blk.layer[0]
            resid, normed = add_norm(sa.layer_norm, resid, delta)
            delta = sa.SelfAttention(normed, position_bias=position_bias, kpad=kpad,
                                     causal=causal, kv_cache=self_cache)
            if self.is_decoder:
                ca = blk.layer[1]
                resid, normed = add_norm(ca.layer_norm, resid, delta)
                delta = ca.EncDecAttention(normed, kv_hidden=enc_hidden,
                                           kpad=enc_kpad, kv_cache=cross_cache,
                                           kv5=kv_layers[i] if kv_layers is not None else None)
            ff = blk.layer[-1]
            resid, normed = add_norm(ff.layer_norm, resid, delta)
            delta = ff.DenseReluDense(normed)
        _, normed = add_norm(self.final_layer_norm, resid, delta)
        return self.dropout(normed)


@dataclass
class Seq2SeqOutput:
    loss: torch.Tensor | None = None
    logits: torch.Tensor | None = None
    encoder_last_hidden_state: torch.Tensor | None = None

    def __getitem__(self, k):
        return getattr(self, k)


def _kpad_from_mask(attention_mask: torch.Tensor | None):
    """(B,S) 1/0 mask -> additive fp32 (B,S) or None when nothing is padded."""
    if attention_mask is None:
        return None
    if bool(attention_mask.all()):
        return None
    return (1.0 - attention_mask.float()) * NEG_INF


class T5ForConditionalGeneration(nn.Module):
    config_class = T5Config

    def __init__(self, config: T5Config):
        super().__init__()
        self.config = config
        self.model_dim = config.d_model
        self.shared = nn.Embedding(config.vocab_size, config.d_model)
        self.encoder = T5Stack(config, self.shared, is_decoder=False)
        self.decoder = T5Stack(config, self.shared, is_decoder=True)
        self.lm_head = nn.Linear(config.d_model, config.vocab_size, bias=False)
        self.init_weights()
        if config.tie_word_embeddings:
            self.lm_head.weight = self.shared.weight

    # -- init (T5 scheme, scaled by initializer_factor) --------------------- #
    def init_weights(self):
        f = self.config.initializer_factor
        d_model, d_kv, d_ff = self.config.d_model, self.config.d_kv, self.config.d_ff
        n_heads = self.config.num_heads
        self.shared.weight.data.normal_(0.0, f * 1.0)
        self.lm_head.weight.data.normal_(0.0, f * d_model**-0.5)
        for m in self.modules():
            if isinstance(m, T5LayerNorm):
                m.weight.data.fill_(f * 1.0)
            elif isinstance(m, T5Attention):
                m.qkv.part(0).data.normal_(0.0, f * (d_model * d_kv) ** -0.5)
                if not m.kv_external:
                    m.qkv.part(1).data.normal_(0.0, f * d_model**-0.5)
                    m.qkv.part(2).data.normal_(0.0, f * d_model**-0.5)
                m.o.weight.data.normal_(0.0, f * (n_heads * d_kv) ** -0.5)
                if m.has_relative_attention_bias:
                    m.relative_attention_bias.weight.data.normal_(0.0, f * d_model**-0.5)
            elif isinstance(m, T5Stack) and m.is_decoder:
                m.cross_kv.data.normal_(0.0, f * d_model**-0.5)
            elif isinstance(m, T5DenseGatedActDense):
                m.wi.weight.data.normal_(0.0, f * d_model**-0.5)
                m.wo.weight.data.normal_(0.0, f * d_ff**-0.5)

    # -- training / scoring forward ----------------------------------------- #
    def _shift_right(self, labels):
        start = self.config.decoder_start_token_id
        shifted = labels.new_zeros(labels.shape)
        shifted[:, 1:] = labels[:, :-1].clone()
        shifted[:, 0] = start
        shifted.masked_fill_(shifted == -100, self.config.pad_token_id)
        return shifted

    @property
    def device(self):
        return self.shared.weight.device

    def forward(self, input_ids=None, attention_mask=None, decoder_input_ids=None,
                decoder_attention_mask=None, labels=None, encoder_outputs=None):
        kpad = _kpad_from_mask(attention_mask)
        if encoder_outputs is None:
            encoder_outputs = self.encoder(input_ids=input_ids, kpad=kpad)
        if decoder_input_ids is None and labels is not None:
            decoder_input_ids = self._shift_right(labels)
        dec = self.decoder(input_ids=decoder_input_ids, kpad=_kpad_from_mask(decoder_attention_mask),
                           enc_hidden=encoder_outputs, enc_kpad=kpad)
        if self.config.tie_word_embeddings:
            dec = dec * (self.model_dim**-0.5)
        # route through ops.linear so the per-shape MFMA GEMM dispatch applies
        # (at B=2/S=512 the lm_head backward-dX runs on the hand-written
        # kernel: 505 vs 291 TF — profiles/r02_gemm_ab.md)
        logits = ops.linear(dec, self.lm_head.weight)
        loss = None
        if labels is not None:
            loss = ops.cross_entropy(logits, labels, ignore_index=-100)
        return Seq2SeqOutput(loss=loss, logits=logits, encoder_last_hidden_state=encoder_outputs)

    # -- greedy generate with preallocated KV cache -------- #
    @torch.no_grad()
    def generate(self, input_ids=None, attention_mask=None, max_new_tokens: int = 128,
                 min_new_tokens: int = 0, eos_token_id=None, use_graph: bool | None = None,
                 **_ignored):
        """Greedy decode. On GPU with a fixed token budget
        (min_new_tokens == max_new_tokens) the whole decode loop is captured
        ONCE as a hipGraph and replayed per call — the loop is launch-bound
        (~90 small kernels per token step) and the replay removes the
        CPU-side gap. Disable with use_graph=False / MI355_GEN_GRAPH=0."""
        if use_graph is None:
            use_graph = os.environ.get("MI355_GEN_GRAPH", "1") != "0"
        if (use_graph and input_ids.is_cuda and min_new_tokens == max_new_tokens
                and not self.training):
            try:
                return self._generate_graphed(input_ids, attention_mask, max_new_tokens,
                                              eos_token_id)
            except Exception as e:
                if not getattr(self, "_gen_graph_warned", False):
                    self._gen_graph_warned = True
                    print(f"# decode-graph capture failed, eager fallback: {e!r}")
        return self._generate_eager(input_ids, attention_mask, max_new_tokens,
                                    min_new_tokens, eos_token_id)

    def _cross_kv_all(self, enc, b):
        """One fused (B*S, n*2*inner) GEMM for every decoder layer's static
        cross-attention kv; returns (n, 2) nested views shaped (B,H,S,D)."""
        cfg = self.config
        h, dkv = cfg.num_heads, cfg.d_kv
        n = len(self.decoder.block)
        kv_all = ops.linear(enc, self.decoder.cross_kv).view(
            b, enc.shape[1], n, 2, h, dkv)
        return [(kv_all[:, :, i, 0].permute(0, 2, 1, 3).contiguous(),
                 kv_all[:, :, i, 1].permute(0, 2, 1, 3).contiguous())
                for i in range(n)]

    def _decode_caches(self, enc, b, max_new_tokens, dev, dt):
        cfg = self.config
        h, dkv = cfg.num_heads, cfg.d_kv
        caches = []
        kv_pairs = self._cross_kv_all(enc, b)
        for i, blk in enumerate(self.decoder.block):
            ck, cv = kv_pairs[i]
            caches.append({
                "self": {"k": torch.zeros(b, h, max_new_tokens, dkv, device=dev, dtype=dt),
                         "v": torch.zeros(b, h, max_new_tokens, dkv, device=dev, dtype=dt),
                         "len": 0},
                "cross": {"k": ck, "v": cv, "static": True},
            })
        return caches

    @torch.no_grad()
    def _generate_graphed(self, input_ids, attention_mask, max_new_tokens,
                          eos_token_id=None):
        cfg = self.config
        b = input_ids.shape[0]
        dev = input_ids.device
        dt = self.shared.weight.dtype
        # The graph path runs only when min_new_tokens == max_new_tokens, and
        # _generate_eager bans EOS at every t < min_new_tokens — so the
        # captured loop must ban EOS at every step too, or fixed-length
        # generations could emit EOS mid-sequence followed by garbage that
        # survives skip_special_tokens decoding.
        eos = cfg.eos_token_id if eos_token_id is None else eos_token_id
        key = (b, input_ids.shape[1], max_new_tokens, str(dev), eos)
        state = getattr(self, "_gen_graph", None)
        kpad = _kpad_from_mask(attention_mask)
        enc = self.encoder(input_ids=input_ids, kpad=kpad)
        if state is None or state["key"] != key or (kpad is None) != (state["kpad"] is None):
            # build static buffers and capture the 128-step decode loop
            caches = self._decode_caches(enc, b, max_new_tokens, dev, dt)
            enc_static = enc.clone()
            kpad_static = kpad.clone() if kpad is not None else None
            out_tokens = torch.zeros(b, max_new_tokens, device=dev, dtype=torch.long)
            attn0 = self.decoder.block[0].layer[0].SelfAttention
            full_bias = attn0.compute_bias(max_new_tokens, max_new_tokens, dev).float()
            bias_steps = [full_bias[:, :, t:t + 1, :t + 1].contiguous()
                          for t in range(max_new_tokens)]
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                tok = torch.full((b, 1), cfg.decoder_start_token_id, device=dev,
                                 dtype=torch.long)
                for i in range(len(caches)):
                    caches[i]["self"]["len"] = 0
                for t in range(max_new_tokens):
                    dec = self._decode_step(tok, enc_static, kpad_static, caches,
                                            t, bias_steps[t].to(dt))
                    if cfg.tie_word_embeddings:
                        dec = dec * (self.model_dim**-0.5)
                    logits = self.lm_head(dec[:, -1])
                    if eos is not None:
                        logits[:, eos] = float("-inf")
                    nxt = logits.argmax(dim=-1)
                    out_tokens[:, t] = nxt
                    tok = nxt.unsqueeze(1)
            state = {"key": key, "graph": g, "enc": enc_static, "kpad": kpad_static,
                     "caches": caches, "out": out_tokens}
            self._gen_graph = state
        # refresh the static inputs, reset self-cache lengths, replay
        state["enc"].copy_(enc)
        if kpad is not None:
            state["kpad"].copy_(kpad)
        kv_pairs = self._cross_kv_all(enc, input_ids.shape[0])
        for i, blk in enumerate(self.decoder.block):
            state["caches"][i]["cross"]["k"].copy_(kv_pairs[i][0])
            state["caches"][i]["cross"]["v"].copy_(kv_pairs[i][1])
            state["caches"][i]["self"]["len"] = 0
        state["graph"].replay()
        return state["out"].clone()

    @torch.no_grad()
    def _generate_eager(self, input_ids=None, attention_mask=None, max_new_tokens: int = 128,
                        min_new_tokens: int = 0, eos_token_id=None, **_ignored):
        cfg = self.config
        eos = cfg.eos_token_id if eos_token_id is None else eos_token_id
        b = input_ids.shape[0]
        dev = input_ids.device
        dt = self.shared.weight.dtype
        kpad = _kpad_from_mask(attention_mask)
        enc = self.encoder(input_ids=input_ids, kpad=kpad)
        caches = self._decode_caches(enc, b, max_new_tokens, dev, dt)
        tokens = torch.full((b, 1), cfg.decoder_start_token_id, device=dev, dtype=torch.long)
        out_tokens = []
        finished = torch.zeros(b, dtype=torch.bool, device=dev)
        attn0 = self.decoder.block[0].layer[0].SelfAttention
        full_bias = attn0.compute_bias(max_new_tokens, max_new_tokens, dev).to(dt)
        for t in range(max_new_tokens):
            # one fp32-contiguous slice per step, shared by all layers (the
            # decode kernel wants f32 bias; avoids 12 per-layer re-copies)
            bias_t = full_bias[:, :, t : t + 1, : t + 1].float().contiguous()
            dec = self._decode_step(tokens, enc, kpad, caches, t, bias_t)
            if cfg.tie_word_embeddings:
                dec = dec * (self.model_dim**-0.5)
            logits = self.lm_head(dec[:, -1])
            if t < min_new_tokens:
                logits[:, eos] = float("-inf")
            nxt = logits.argmax(dim=-1)
            nxt = torch.where(finished, torch.full_like(nxt, cfg.pad_token_id), nxt)
            out_tokens.append(nxt)
            finished |= nxt == eos
            if bool(finished.all()):
                break
            tokens = nxt.unsqueeze(1)
        return torch.stack(out_tokens, dim=1)

    def _decode_step(self, tokens, enc, kpad, caches, t, bias_t):
        x = self.decoder.dropout(
            ops.token_embedding(tokens, self.decoder.embed_tokens.weight))
        dec = self.decoder
        resid, delta = x, None

        def add_norm(ln, resid, delta):
            if delta is None:
                return resid, ln(resid)
            return ops.add_rms_norm(resid, delta, ln.weight, ln.variance_epsilon)

        for i, blk in enumerate(dec.block):
            sa, ca, ff = blk.layer[0], blk.layer[1], blk.layer[2]
            resid, normed = add_norm(sa.layer_norm, resid, delta)
            delta = sa.SelfAttention(normed, position_bias=bias_t,
                                     kv_cache=caches[i]["self"])
            resid, normed = add_norm(ca.layer_norm, resid, delta)
            delta = ca.EncDecAttention(normed, kv_hidden=enc, kpad=kpad,
                                       kv_cache=caches[i]["cross"])
            resid, normed = add_norm(ff.layer_norm, resid, delta)
            delta = ff.DenseReluDense(normed)
        _, normed = add_norm(dec.final_layer_norm, resid, delta)
        return normed

    # -- HF-layout checkpoint I/O --------------------------- #
    def save_pretrained(self, save_dir: str, safe_serialization: bool = True):
        os.makedirs(save_dir, exist_ok=True)
        self.config.save_pretrained(save_dir)
        state = {k: v for k, v in self.state_dict().items()}
        # encoder/decoder embed_tokens alias the shared embedding (HF layout
        # stores only "shared.weight"); a tied lm_head is also dropped.
        state.pop("encoder.embed_tokens.weight", None)
        state.pop("decoder.embed_tokens.weight", None)
        if self.config.tie_word_embeddings:
            state.pop("lm_head.weight", None)
        if safe_serialization:
            from safetensors.torch import save_file

            save_file({k: v.contiguous().cpu() for k, v in state.items()},
                      os.path.join(save_dir, "model.safetensors"),
                      metadata={"format": "pt"})
        else:
            torch.save(state, os.path.join(save_dir, "pytorch_model.bin"))

    @classmethod
    def from_pretrained(cls, path: str, torch_dtype=None, config: T5Config | None = None,
                        **_ignored):
        if config is None and os.path.isdir(path):
            config = T5Config.from_pretrained(path)
        elif config is None:
            config = t5_config(path)
        model = cls(config)
        st_file = os.path.join(path, "model.safetensors") if os.path.isdir(path) else None
        if st_file and os.path.exists(st_file):
            from safetensors.torch import load_file

            state = load_file(st_file)
            missing, unexpected = model.load_state_dict(state, strict=False)
            aliases = {"encoder.embed_tokens.weight", "decoder.embed_tokens.weight"}
            if model.config.tie_word_embeddings:
                aliases.add("lm_head.weight")
            missing = [k for k in missing if k not in aliases]
            if missing:
                raise RuntimeError(f"missing keys in checkpoint: {missing}")
            if model.config.tie_word_embeddings:
                model.lm_head.weight = model.shared.weight
        elif os.path.isdir(path) and os.path.exists(os.path.join(path, "pytorch_model.bin")):
            state = torch.load(os.path.join(path, "pytorch_model.bin"), map_location="cpu",
                               weights_only=True)
            model.load_state_dict(state, strict=False)
        if torch_dtype is not None:
            model = model.to(torch_dtype)
        return model
