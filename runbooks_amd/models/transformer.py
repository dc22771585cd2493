"""TP-aware decoder-only transformer covering the llama / falcon / OPT
families the reference platform serves.

One implementation, three family knobs (ModelConfig):
  * norm: RMSNorm (fused gfx950 kernel) or LayerNorm
  * act: SwiGLU (llama), GELU (falcon), ReLU (OPT)
  * pos: RoPE (fused kernel) or learned embeddings (OPT)
  * parallel_residual: falcon's attn+mlp-in-parallel block
  * GQA/MQA via num_kv_heads (llama2-70b 8, falcon-40b 8, falcon-7b 1)

Three forward modes:
  * ``forward(tokens)``             — training, full causal attention
  * ``prefill(...)``                — fills the paged KV cache, returns
                                      last-position logits
  * ``decode(...)``                 — one token per sequence against the
                                      paged cache (gfx950 decode kernel)

TP: attention QKV is column-parallel (head-sharded), the output
projection row-parallel (one xGMI all-reduce); MLP up column / down row
(second all-reduce). KV heads shard when divisible by tp, otherwise the
KV projection is replicated (MQA at tp>1) and each rank keeps a full KV
cache — 288 GB HBM per GPU makes that the right trade on MI355X.
"""
from __future__ import annotations

import math

import torch
from torch import nn

from .. import ops
from ..parallel import (
    ColumnParallelLinear,
    RowParallelLinear,
    comm,
    gather_from_tp,
)
from .config import ModelConfig


def _norm_module(cfg: ModelConfig, dtype):
    if cfg.norm == "rmsnorm":
        return ops.RMSNorm(cfg.hidden_size, eps=cfg.norm_eps, dtype=dtype)
    return nn.LayerNorm(cfg.hidden_size, eps=cfg.norm_eps, dtype=dtype)


class Attention(nn.Module):
    def __init__(self, cfg: ModelConfig, dtype, tp: int):
        super().__init__()
        self.cfg = cfg
        self.tp = tp
        self.dh = cfg.head_dim
        assert cfg.num_heads % tp == 0, "num_heads must divide tp"
        self.hq = cfg.num_heads // tp
        self.kv_sharded = cfg.num_kv_heads % tp == 0
        self.hkv = cfg.num_kv_heads // tp if self.kv_sharded else cfg.num_kv_heads
        self.scale = 1.0 / math.sqrt(self.dh)

        q_out = cfg.num_heads * self.dh
        kv_out = cfg.num_kv_heads * self.dh
        # qwen2-style models carry bias on q/k/v but not on o_proj
        in_bias = cfg.attn_bias or cfg.qkv_bias
        self.q_proj = ColumnParallelLinear(cfg.hidden_size, q_out,
                                           bias=in_bias, tp_size=tp, dtype=dtype)
        if self.kv_sharded:
            self.k_proj = ColumnParallelLinear(cfg.hidden_size, kv_out,
                                               bias=in_bias, tp_size=tp,
                                               dtype=dtype, gather_input=False)
            self.v_proj = ColumnParallelLinear(cfg.hidden_size, kv_out,
                                               bias=in_bias, tp_size=tp,
                                               dtype=dtype, gather_input=False)
        else:  # replicate KV (MQA with tp > num_kv_heads)
            self.k_proj = ops.Linear(cfg.hidden_size, kv_out, bias=in_bias,
                                    dtype=dtype)
            self.v_proj = ops.Linear(cfg.hidden_size, kv_out, bias=in_bias,
                                    dtype=dtype)
        self.o_proj = RowParallelLinear(cfg.num_heads * self.dh, cfg.hidden_size,
                                        bias=cfg.attn_bias, tp_size=tp, dtype=dtype)

    def _qkv(self, x, cos, sin, positions):
        # x: [T, hidden] token-major
        T = x.shape[0]
        if getattr(self, "_qkv_w", None) is not None and \
                not torch.is_grad_enabled():
            # fused inference path (fuse_for_inference): one wide GEMM
            # instead of three skinny ones
            qo = self.hq * self.dh
            kvo = self.hkv * self.dh
            y = ops.fast_linear(x, self._qkv_w)
            q = y[:, :qo].contiguous().view(T, self.hq, self.dh)
            k = y[:, qo:qo + kvo].contiguous().view(T, self.hkv, self.dh)
            v = y[:, qo + kvo:].contiguous().view(T, self.hkv, self.dh)
        else:
            q = self.q_proj(x).view(T, self.hq, self.dh)
            k = self.k_proj(x).view(T, self.hkv, self.dh)
            v = self.v_proj(x).view(T, self.hkv, self.dh)
        if self.cfg.pos == "rope":
            q = ops.rope(q, cos, sin, positions)
            k = ops.rope(k, cos, sin, positions)
        return q, k, v

    def forward_train(self, x, cos, sin, positions, B, S):
        q, k, v = self._qkv(x, cos, sin, positions)
        q = q.view(B, S, self.hq, self.dh)
        k = k.view(B, S, self.hkv, self.dh)
        v = v.view(B, S, self.hkv, self.dh)
        o = ops.causal_attention(q, k, v, scale=self.scale)
        return self.o_proj(o.reshape(B * S, self.hq * self.dh))

    def forward_prefill(self, x, cos, sin, positions, kc, vc, slot_mapping, B, S):
        q, k, v = self._qkv(x, cos, sin, positions)
        ops.kv_append(k, v, kc, vc, slot_mapping)
        q = q.view(B, S, self.hq, self.dh)
        o = ops.flash_prefill(q, k.view(B, S, self.hkv, self.dh),
                              v.view(B, S, self.hkv, self.dh), scale=self.scale)
        return self.o_proj(o.reshape(B * S, self.hq * self.dh))

    def forward_decode(self, x, cos, sin, positions, kc, vc, slot_mapping,
                       block_tables, seq_lens, nsplit=None, slab_ok=False,
                       seq_starts=None):
        B = x.shape[0]
        if getattr(self, "_qkv_w", None) is not None and \
                self.cfg.pos == "rope" and ops.use_hip(x) and \
                x.dtype == torch.bfloat16 and kc.dtype == torch.bfloat16:
            # fused decode hot path: one wide GEMM, then one kernel doing
            # rope(q), rope(k)->cache, v->cache (csrc/qkv_fused.hip).
            # fp8 caches take the generic path below (rope + fp8-quant
            # kv_append): that mode targets long-context KV-bound decode
            # where the attention read, not the epilogue, dominates.
            y = ops.fast_linear(x, self._qkv_w)
            q = ops.ext().qkv_rope_append(y, cos, sin, positions, kc, vc,
                                          slot_mapping, self.hq)
        else:
            q, k, v = self._qkv(x, cos, sin, positions)
            ops.kv_append(k, v, kc, vc, slot_mapping)
        o, oswz = ops.paged_decode_with_operand(
            q, kc, vc, block_tables, seq_lens, scale=self.scale,
            nsplit=nsplit, seq_starts=seq_starts)
        o2 = o.reshape(B, self.hq * self.dh)
        if oswz is not None:
            o2._rb_swz = oswz
        if slab_ok and self.tp == 1 and self.o_proj.bias is None:
            out, _ = ops.decode_linear_raw(o2, self.o_proj.weight)
            return out
        return self.o_proj(o2)


class MLP(nn.Module):
    def __init__(self, cfg: ModelConfig, dtype, tp: int):
        super().__init__()
        self.act = cfg.act
        if cfg.act in ("silu_glu", "gelu_glu"):
            self.gate_proj = ColumnParallelLinear(cfg.hidden_size, cfg.intermediate_size,
                                                  bias=cfg.mlp_bias, tp_size=tp, dtype=dtype)
            self.up_proj = ColumnParallelLinear(cfg.hidden_size, cfg.intermediate_size,
                                                bias=cfg.mlp_bias, tp_size=tp, dtype=dtype,
                                                gather_input=False)
        else:
            self.fc1 = ColumnParallelLinear(cfg.hidden_size, cfg.intermediate_size,
                                            bias=cfg.mlp_bias, tp_size=tp, dtype=dtype)
        self.down_proj = RowParallelLinear(cfg.intermediate_size, cfg.hidden_size,
                                           bias=cfg.mlp_bias, tp_size=tp, dtype=dtype)

    def forward(self, x, slab_ok=False):
        if self.act == "silu_glu":
            if getattr(self, "_gateup_w", None) is not None and \
                    not torch.is_grad_enabled():
                y = ops.fast_linear(x, self._gateup_w)
                if ops.use_hip(y):
                    I = y.shape[-1] // 2
                    if (y.dtype == torch.bfloat16 and I % 16 == 0 and
                            y.numel() // y.shape[-1] <= 32):
                        h, swz = ops.ext().swiglu_packed_dec(y)
                        h._rb_swz = swz
                        if slab_ok and self.down_proj.bias is None:
                            out, _ = ops.decode_linear_raw(
                                h, self.down_proj.weight)
                            return out
                        return self.down_proj(h)
                    return self.down_proj(ops.ext().swiglu_packed(y))
                half = y.shape[-1] // 2
                return self.down_proj(ops.swiglu(
                    y[:, :half].contiguous(), y[:, half:].contiguous()))
            return self.down_proj(ops.swiglu(self.gate_proj(x), self.up_proj(x)))
        if self.act == "gelu_glu":
            # gemma GeGLU (tanh approximation); fused gate/up GEMM +
            # csrc geglu_packed epilogue (GPU-validated r2).
            if getattr(self, "_gateup_w", None) is not None and \
                    not torch.is_grad_enabled():
                y = ops.fast_linear(x, self._gateup_w)
                if ops.use_hip(y):
                    I = y.shape[-1] // 2
                    if (y.dtype == torch.bfloat16 and I % 16 == 0 and
                            y.numel() // y.shape[-1] <= 32):
                        h, swz = ops.ext().geglu_packed_dec(y)
                        h._rb_swz = swz
                        if slab_ok and self.down_proj.bias is None:
                            out, _ = ops.decode_linear_raw(
                                h, self.down_proj.weight)
                            return out
                        return self.down_proj(h)
                    return self.down_proj(ops.ext().geglu_packed(y))
                half = y.shape[-1] // 2
                return self.down_proj(
                    torch.nn.functional.gelu(y[:, :half],
                                             approximate="tanh") *
                    y[:, half:])
            return self.down_proj(
                torch.nn.functional.gelu(self.gate_proj(x),
                                         approximate="tanh") * self.up_proj(x))
        h = self.fc1(x)
        if self.act == "gelu":
            h = torch.nn.functional.gelu(h)
        elif self.act == "gelu_tanh":      # gpt2 "gelu_new"
            h = torch.nn.functional.gelu(h, approximate="tanh")
        else:
            h = torch.relu(h)
        return self.down_proj(h)


class Block(nn.Module):
    def __init__(self, cfg: ModelConfig, dtype, tp: int):
        super().__init__()
        self.cfg = cfg
        self.attn = Attention(cfg, dtype, tp)
        self.mlp = MLP(cfg, dtype, tp)
        self.norm1 = _norm_module(cfg, dtype)
        self.norm2 = None if cfg.single_norm else _norm_module(cfg, dtype)

    def forward(self, x, attn_fn):
        """attn_fn(h) runs the right attention mode on normed input."""
        if self.cfg.parallel_residual:
            if self.cfg.single_norm:
                # falcon-7b: one shared ln feeds both branches
                h = self.norm1(x)
                return x + attn_fn(h) + self.mlp(h)
            # falcon-40b: x + attn(ln_attn(x)) + mlp(ln_mlp(x))
            return x + attn_fn(self.norm1(x)) + self.mlp(self.norm2(x))
        x = x + attn_fn(self.norm1(x))
        return x + self.mlp(self.norm2(x))

    def forward_decode_fused(self, x_base, delta, attn_fn):
        """Decode hot path (sequential-residual RMSNorm models): the
        residual add is fused into the next norm's kernel
        (rmsnorm_res_fwd_dec), so the layer's residual chain costs zero
        standalone elementwise launches. Returns (new_base, new_delta)
        with the true hidden state = new_base + new_delta (the caller
        folds the final pending delta into norm_f)."""
        ext = ops.ext()
        eps = self.norm1.eps

        def res_norm(base, d, w, e):
            if d.dtype == torch.float32:      # uncombined split-K slab
                return ext.rmsnorm_res_slab_fwd_dec(base, d, w, e)
            return ext.rmsnorm_res_fwd_dec(base, d, w, e)

        if delta is None:
            y1 = self.norm1(x_base)          # attaches _rb_swz itself
            xr1 = x_base
        else:
            xr1, y1, s1 = res_norm(x_base, delta, self.norm1.weight, eps)
            y1._rb_swz = s1
        a = attn_fn(y1)
        xr2, y2, s2 = res_norm(xr1, a, self.norm2.weight, self.norm2.eps)
        y2._rb_swz = s2
        return xr2, self.mlp(y2, slab_ok=True)


class Transformer(nn.Module):
    def __init__(self, cfg: ModelConfig, dtype=torch.bfloat16,
                 tp: int | None = None, device=None):
        super().__init__()
        self.cfg = cfg
        self.tp = tp if tp is not None else comm.world_size()
        self.dtype = dtype
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size, dtype=dtype)
        if cfg.pos == "learned":
            self.embed_pos = nn.Embedding(cfg.max_seq_len, cfg.hidden_size, dtype=dtype)
        self.blocks = nn.ModuleList(Block(cfg, dtype, self.tp)
                                    for _ in range(cfg.num_layers))
        self.norm_f = _norm_module(cfg, dtype)
        # vocab-parallel head at TP>1 (untied vocab divisible by tp):
        # each rank's head GEMM shrinks tp-fold; full logits reassemble
        # with one all-gather over xGMI (B x V/tp bf16 per rank, cheap
        # next to the 8x GEMM saving on a 32k-152k vocab).
        self._lm_vocab_parallel = (self.tp > 1 and
                                   cfg.vocab_size % self.tp == 0 and
                                   not cfg.tie_embeddings)
        if self._lm_vocab_parallel:
            self.lm_head = ColumnParallelLinear(
                cfg.hidden_size, cfg.vocab_size, bias=False,
                tp_size=self.tp, dtype=dtype)
        else:
            self.lm_head = ops.Linear(cfg.hidden_size, cfg.vocab_size,
                                      bias=False, dtype=dtype)
        if cfg.tie_embeddings:
            self.lm_head.weight = self.embed.weight
        if cfg.pos == "rope":
            cos, sin = ops.rope_tables(cfg.head_dim, cfg.max_seq_len, cfg.rope_theta)
            self.register_buffer("rope_cos", cos, persistent=False)
            self.register_buffer("rope_sin", sin, persistent=False)
        else:
            self.rope_cos = self.rope_sin = None
        self.reset_parameters()
        if device is not None:
            self.to(device)

    def reset_parameters(self):
        for m in self.modules():
            if isinstance(m, (nn.Linear, ColumnParallelLinear, RowParallelLinear)):  # ops.Linear subclasses nn.Linear
                nn.init.normal_(m.weight, std=0.02)
                if getattr(m, "bias", None) is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=0.02)

    # -- shared ----------------------------------------------------------------
    def _embed(self, tokens, positions):
        """Token (+learned position) embedding; gemma scales by
        sqrt(hidden) (cfg.embed_scale, 1.0 elsewhere)."""
        x = self.embed(tokens.reshape(-1))
        if self.cfg.embed_scale != 1.0:
            x = x * self.cfg.embed_scale
        if self.cfg.pos == "learned":
            x = x + self.embed_pos(positions.long())
        return x

    # -- training --------------------------------------------------------------
    def forward(self, tokens: torch.Tensor) -> torch.Tensor:
        """tokens [B, S] -> logits [B, S, V] (training / eval, no cache)."""
        B, S = tokens.shape
        positions = torch.arange(S, device=tokens.device, dtype=torch.int32)
        positions = positions.repeat(B)
        x = self._embed(tokens, positions)
        ckpt = getattr(self, "grad_checkpointing", False) and \
            torch.is_grad_enabled()
        for blk in self.blocks:
            fn = lambda h, b=blk: b.attn.forward_train(  # noqa: E731
                h, self.rope_cos, self.rope_sin, positions, B, S)
            if ckpt:
                # recompute this block's activations in backward: peak
                # activation memory drops from O(layers) to O(1) blocks
                # (full fine-tunes of 70B-class models at long seq; LoRA
                # at default shapes fits 288 GB without it)
                x = torch.utils.checkpoint.checkpoint(
                    blk, x, fn, use_reentrant=False)
            else:
                x = blk(x, fn)
        x = self.norm_f(x)
        logits = self.lm_head(x)
        if self._lm_vocab_parallel:
            logits = gather_from_tp(logits)
        return logits.view(B, S, -1)

    def enable_grad_checkpointing(self, enabled: bool = True) -> None:
        self.grad_checkpointing = enabled

    # -- serving ---------------------------------------------------------------
    @torch.no_grad()
    def prefill(self, tokens, positions, caches, slot_mapping):
        """tokens [B, S]; caches: list of (k_cache, v_cache) per layer.
        Returns last-position logits [B, V]."""
        B, S = tokens.shape
        x = self._embed(tokens, positions)
        for blk, (kc, vc) in zip(self.blocks, caches):
            fn = lambda h, b=blk, kc=kc, vc=vc: b.attn.forward_prefill(  # noqa: E731
                h, self.rope_cos, self.rope_sin, positions, kc, vc, slot_mapping, B, S)
            x = blk(x, fn)
        x = self.norm_f(x.view(B, S, -1)[:, -1])
        logits = self.lm_head(x)
        return gather_from_tp(logits) if self._lm_vocab_parallel else logits

    @torch.no_grad()
    def decode(self, tokens, positions, caches, slot_mapping, block_tables,
               seq_lens, nsplit=None, seq_starts=None):
        """tokens [B] (one new token per sequence) -> logits [B, V].
        ``nsplit`` fixes the paged-decode work split (required under
        hipGraph capture, serve/graph.py)."""
        B = tokens.shape[0]
        x = self._embed(tokens.view(B, 1), positions)
        import os as _os
        if (ops.use_hip(x) and x.dtype == torch.bfloat16 and B <= 32 and
                self.cfg.norm == "rmsnorm" and
                not self.cfg.parallel_residual and not self.cfg.single_norm
                and self.cfg.hidden_size % 16 == 0 and
                _os.environ.get("RB_FUSED_RESID", "1") == "1"):
            # fused residual chain: zero standalone adds in the hot loop
            delta = None
            for blk, (kc, vc) in zip(self.blocks, caches):
                fn = lambda h, b=blk, kc=kc, vc=vc: b.attn.forward_decode(  # noqa: E731
                    h, self.rope_cos, self.rope_sin, positions, kc, vc,
                    slot_mapping, block_tables, seq_lens, nsplit=nsplit,
                    slab_ok=True, seq_starts=seq_starts)
                x, delta = blk.forward_decode_fused(x, delta, fn)
            if delta.dtype == torch.float32:
                _, y, swz = ops.ext().rmsnorm_res_slab_fwd_dec(
                    x, delta, self.norm_f.weight, self.norm_f.eps)
            else:
                _, y, swz = ops.ext().rmsnorm_res_fwd_dec(
                    x, delta, self.norm_f.weight, self.norm_f.eps)
            y._rb_swz = swz
            logits = self.lm_head(y)
            return (gather_from_tp(logits) if self._lm_vocab_parallel
                    else logits)
        for blk, (kc, vc) in zip(self.blocks, caches):
            fn = lambda h, b=blk, kc=kc, vc=vc: b.attn.forward_decode(  # noqa: E731
                h, self.rope_cos, self.rope_sin, positions, kc, vc, slot_mapping,
                block_tables, seq_lens, nsplit=nsplit, seq_starts=seq_starts)
            x = blk(x, fn)
        x = self.norm_f(x)
        logits = self.lm_head(x)
        return gather_from_tp(logits) if self._lm_vocab_parallel else logits

    # -- cache helpers ----------------------------------------------------------
    def local_kv_heads(self) -> int:
        return self.blocks[0].attn.hkv

    def alloc_caches(self, num_blocks: int, device, fp8: bool = False):
        # Transposed-V blocks feed the MFMA decode kernel's direct V^T
        # fragment reads (GQA/MQA models, bf16, Dh<=128). The cache
        # shape is the routing signal all the way down (kv_append,
        # paged_decode). RB_DECODE_MFMA=0 reverts to the scalar path.
        import os
        attn = self.blocks[0].attn
        vt = ((fp8 or self.dtype == torch.bfloat16)
              and attn.hq // attn.hkv >= 4 and self.cfg.head_dim <= 128
              and ops.BLOCK_SIZE == 16
              and os.environ.get("RB_DECODE_MFMA", "1") != "0")
        return [ops.alloc_kv_cache(num_blocks, self.local_kv_heads(),
                                   self.cfg.head_dim, device,
                                   dtype=self.dtype, fp8=fp8,
                                   v_transposed=vt)
                for _ in range(self.cfg.num_layers)]


def fuse_for_inference(model: "Transformer",
                       load_in_8bit: bool = False) -> "Transformer":
    """Fuse each block's QKV and gate/up weights into single tensors for
    serving: one wide GEMM per group instead of 2-3 skinny ones (the
    batch-<=32 decode GEMMs are weight-bandwidth bound, and hipBLASLt is
    markedly more efficient at the fused N — profiles/). The original
    parameters become views into the fused tensor, so no extra HBM and
    the separate-projection training path keeps working.
    """
    for blk in model.blocks:
        attn = blk.attn
        if (attn.q_proj.bias is None and
                getattr(attn.k_proj, "bias", None) is None and
                attn.q_proj.weight.shape[1] == attn.k_proj.weight.shape[1]):
            fused = torch.cat([attn.q_proj.weight.data,
                               attn.k_proj.weight.data,
                               attn.v_proj.weight.data], dim=0).contiguous()
            qo = attn.q_proj.weight.shape[0]
            kvo = attn.k_proj.weight.shape[0]
            attn._qkv_w = fused
            attn.q_proj.weight = nn.Parameter(fused[:qo],
                                              requires_grad=False)
            attn.k_proj.weight = nn.Parameter(fused[qo:qo + kvo],
                                              requires_grad=False)
            attn.v_proj.weight = nn.Parameter(fused[qo + kvo:],
                                              requires_grad=False)
        mlp = blk.mlp
        import os as _os
        # geglu_packed GPU-validated (GPUTEST r2: vs fp32 tanh-gelu ref);
        # default-on, RB_FUSED_GEGLU=0 reverts to eager
        glu_fusable = model.cfg.act == "silu_glu" or (
            model.cfg.act == "gelu_glu" and
            _os.environ.get("RB_FUSED_GEGLU", "1") == "1")
        if glu_fusable and mlp.gate_proj.bias is None:
            fused = torch.cat([mlp.gate_proj.weight.data,
                               mlp.up_proj.weight.data],
                              dim=0).contiguous()
            half = mlp.gate_proj.weight.shape[0]
            mlp._gateup_w = fused
            mlp.gate_proj.weight = nn.Parameter(fused[:half],
                                                requires_grad=False)
            mlp.up_proj.weight = nn.Parameter(fused[half:],
                                              requires_grad=False)

    if not load_in_8bit and torch.cuda.is_available() and ops.has_hip() and \
            _os.environ.get("RB_DECODE_GEMM", "1") == "1":
        # Register fragment-lane-major weight copies for the v2 decode
        # GEMM (csrc/decode_gemm.hip): doubles decode-weight memory, which
        # 288 GB HBM3E affords for every BASELINE config at TP=1 except
        # llama2-70b (which serves via MODEL_LOAD_IN_8BIT / TP>1). Budget
        # guard: keep at least ~35% of the GPU free for KV cache.
        from ..ops.linear import register_decode_weight
        ext = ops.ext()
        cands = []
        for blk in model.blocks:
            for w in (getattr(blk.attn, "_qkv_w", None),
                      getattr(blk.mlp, "_gateup_w", None),
                      blk.attn.o_proj.weight,
                      blk.mlp.down_proj.weight):
                if w is not None and ext.decode_gemm_supported(
                        32, w.shape[0], w.shape[1]):
                    cands.append(w)
        if model.lm_head.weight.shape[0] % 32 == 0 and \
                ext.decode_gemm_supported(32, model.lm_head.weight.shape[0],
                                          model.lm_head.weight.shape[1]):
            cands.append(model.lm_head.weight)
        need = sum(2 * w.numel() for w in cands)
        free, total = torch.cuda.mem_get_info()
        if free - need >= int(0.35 * total):
            for w in cands:
                register_decode_weight(w)

    if load_in_8bit and torch.cuda.is_available():
        # MODEL_LOAD_IN_8BIT: decode weights additionally stored as OCP
        # e4m3 + per-channel scales — the decode GEMM streams half the
        # bytes (ops/csrc/skinny_gemm.hip fp8 path). bf16 masters stay for
        # prefill/big-M GEMMs: 288 GB HBM3E makes both-copies the right
        # trade on MI355X.
        from ..ops.linear import quantize_fp8
        for blk in model.blocks:
            for w in (getattr(blk.attn, "_qkv_w", None),
                      getattr(blk.mlp, "_gateup_w", None),
                      blk.attn.o_proj.weight,
                      blk.mlp.down_proj.weight):
                if w is not None and w.shape[0] % 64 == 0 and \
                        w.shape[1] % 256 == 0:
                    quantize_fp8(w)
        if model.lm_head.weight.shape[0] % 64 == 0 and \
                model.lm_head.weight.shape[1] % 256 == 0:
            quantize_fp8(model.lm_head.weight)
    return model


def build_model(name_or_cfg, dtype=torch.bfloat16, tp: int | None = None,
                device=None, seed: int = 0) -> Transformer:
    from .config import get_config
    cfg = name_or_cfg if isinstance(name_or_cfg, ModelConfig) else get_config(name_or_cfg)
    torch.manual_seed(seed)
    if device is not None:
        # construct directly on the target device — a 7B bf16 random init
        # takes minutes on CPU but is instant on the GPU, and 8 DP ranks
        # would otherwise all churn host RAM at once
        with torch.device(device):
            return Transformer(cfg, dtype=dtype, tp=tp)
    return Transformer(cfg, dtype=dtype, tp=tp)
