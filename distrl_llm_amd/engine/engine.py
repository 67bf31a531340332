"""Paged-KV generation engine: prefill + continuous-batching decode.

Native replacement for the reference's in-process vLLM engine
(`policy.fast_generate`, reference distributed_actor.py:147-172): batched
sampling with a paged KV cache, n-candidate fan-out per prompt (prompt KV
blocks shared across the n candidates via refcounting — prefill runs once
per prompt, not n times), per-call SamplingParams, and EOS/max-token
termination. Decode is the hot loop; on GPU every custom
op routes through the gfx950 HIP extension via ops.functional and the
step can be hipGraph-captured (engine.graph).

The engine reads the SAME weight/LoRA tensors the learner trains, so
in-process weight sync is free and cross-process sync is one RCCL
broadcast into these tensors.
"""

from __future__ import annotations

import os
from collections import OrderedDict
from typing import List, Optional

import torch
import torch.nn.functional as F

from ..config import EngineConfig, SamplingParams
from ..models.model import CausalLM
from ..ops import functional as OF
from ..ops import reference as R
from .kvcache import KVCachePool, Sequence
from ..utils.trace import trace_range


class Engine:
    def __init__(self, model: CausalLM, cfg: EngineConfig,
                 device: Optional[torch.device] = None, seed: int = 0):
        self.model = model
        self.cfg = cfg
        self.spec = model.spec
        self.device = device if device is not None else model.device
        self.dtype = model.dtype_
        self.scale = self.spec.head_dim ** -0.5
        self._seq_counter = 0
        self._session_cache = None  # DISTRL_GRAPH_CACHE=1 lazily creates
        # automatic prefix caching (cfg.enable_prefix_caching): LRU map
        # {tuple(prompt[:k*block_size]): block_id}; the cache holds its
        # own refcount on every entry so blocks survive sequence finish
        self._prefix_cache = (OrderedDict()
                              if cfg.enable_prefix_caching else None)
        self._prefix_hits = 0   # blocks reused (tests/metrics)
        self._prefix_evicts = 0
        self.generator = torch.Generator(device=self.device)
        self.generator.manual_seed(seed)

        d = self.spec.head_dim
        self._inv_freq = 1.0 / (self.spec.rope_theta ** (
            torch.arange(0, d, 2, device=self.device, dtype=torch.float32) / d))

        num_blocks = cfg.num_kv_blocks
        if num_blocks <= 0:
            num_blocks = self._derive_num_blocks()
        self.pool = KVCachePool(self.spec.num_layers, num_blocks,
                                cfg.kv_block_size, self.spec.num_kv_heads,
                                self.spec.head_dim, self.dtype, self.device)
        if self.device.type == "cuda":
            from .weights import FusedWeights
            self.fused = FusedWeights(model)
        else:
            self.fused = None

    # ------------------------------------------------------------ sizing

    def _derive_num_blocks(self) -> int:
        """Size the KV pool from free HBM (re-derived for 288 GB per GPU,
        not copied from the reference's 24 GB fractions — SURVEY.md
        §2.6-10)."""
        s = self.spec
        bs = self.cfg.kv_block_size
        per_block = KVCachePool.pool_size_bytes(1, s.num_layers, bs,
                                                s.num_kv_heads, s.head_dim,
                                                self.dtype)
        if self.device.type == "cuda":
            free, _total = torch.cuda.mem_get_info(self.device)
            budget = int(free * self.cfg.gpu_memory_utilization)
        else:
            budget = 64 * 1024 * 1024  # CPU tests: 64 MB pool
        n = max(budget // per_block, 16)
        max_needed = self.cfg.max_num_seqs * KVCachePool.blocks_for(
            self.cfg.max_seq_length, bs)
        return int(min(n, max_needed))

    # ---------------------------------------------------------- weights

    def _proj(self, mod, x):
        """Projection through a LoRALinear (base GEMM + adapter)."""
        return mod(x)

    # ---------------------------------------------------------- prefill

    @torch.no_grad()
    def _prefill_batch(self, seqs: List[Sequence]) -> torch.Tensor:
        """Run the prompt phase for a batch of fresh sequences (one per
        prompt; fan-out happens after). Writes prompt KV into the pool and
        returns last-token logits (len(seqs), V).

        With ``enable_prefix_caching`` on, sequences whose prompt starts
        with already-cached full blocks reuse those blocks (incref, no
        recompute) and prefill only the tail — through the DECODE step,
        so no new attention kernel is involved (classic serving win: a
        long shared system prompt + short per-user tails). Full blocks of
        every prefilled prompt are then inserted into the LRU cache."""
        if self._prefix_cache is None:
            return self._prefill_flat(seqs)
        bs = self.pool.block_size
        hits, misses, logits = [], [], {}
        for q in seqs:
            ids = q.prompt_ids
            blocks = []
            # cap at (L-1)//bs blocks so >= 1 tail token yields logits
            for i in range((len(ids) - 1) // bs):
                key = tuple(ids[:(i + 1) * bs])
                blk = self._prefix_cache.get(key)
                if blk is None:
                    break
                blocks.append(blk)
                self._prefix_cache.move_to_end(key)
            if blocks:
                for b in blocks:
                    self.pool.allocator.incref(b)
                q.block_table = list(blocks)
                q.context_len = len(blocks) * bs
                self._prefix_hits += len(blocks)
                hits.append(q)
            else:
                misses.append(q)
        if misses:
            lg = self._prefill_flat(misses)
            for i, q in enumerate(misses):
                logits[id(q)] = lg[i:i + 1]
        if hits:
            lg = self._prefill_tail(hits)
            for i, q in enumerate(hits):
                logits[id(q)] = lg[i:i + 1]
        self._prefix_insert(seqs)
        return torch.cat([logits[id(q)] for q in seqs], 0)

    def _prefill_tail(self, seqs: List[Sequence]) -> torch.Tensor:
        """Teacher-force the uncached prompt tail of cache-hit sequences
        through the decode step (KV scatter + paged attention against the
        cached context — existing, GPU-validated kernels only). Returns
        last-token logits per sequence. Sequential over the longest tail;
        the cache trades that for skipping the cached prefix entirely."""
        device = self.device
        bs = self.pool.block_size
        out: List[Optional[torch.Tensor]] = [None] * len(seqs)
        active = list(range(len(seqs)))
        while active:
            tokens, positions, slots, ctx = [], [], [], []
            for i in active:
                q = seqs[i]
                pos = q.context_len
                tokens.append(q.prompt_ids[pos])
                positions.append(pos)
                bi = pos // bs
                if bi == len(q.block_table):
                    q.block_table.append(self.pool.allocator.alloc())
                slots.append(q.block_table[bi] * bs + pos % bs)
                ctx.append(pos + 1)
            max_nb = max(len(seqs[i].block_table) for i in active)
            block_tables = torch.zeros(len(active), max_nb,
                                       dtype=torch.int32, device=device)
            for row, i in enumerate(active):
                bt = seqs[i].block_table
                block_tables[row, :len(bt)] = torch.tensor(
                    bt, dtype=torch.int32)
            lg = self._decode_forward(
                torch.tensor(tokens, dtype=torch.long, device=device),
                torch.tensor(positions, dtype=torch.long, device=device),
                torch.tensor(slots, dtype=torch.long, device=device),
                block_tables,
                torch.tensor(ctx, dtype=torch.int32, device=device))
            nxt = []
            for row, i in enumerate(active):
                q = seqs[i]
                q.context_len += 1
                if q.context_len >= len(q.prompt_ids):
                    out[i] = lg[row:row + 1]
                else:
                    nxt.append(i)
            active = nxt
        return torch.cat([o for o in out], 0)

    def _prefix_insert(self, seqs: List[Sequence]) -> None:
        """Insert every fully-written prompt block into the LRU cache
        (the cache takes its own refcount, so blocks outlive their
        sequences). Duplicate keys keep the first block. Note: evicting
        a shorter-prefix entry orphans its longer continuations until
        LRU reclaims them — lookups walk prefixes in order."""
        bs = self.pool.block_size
        for q in seqs:
            ids = q.prompt_ids
            for i in range(len(ids) // bs):
                key = tuple(ids[:(i + 1) * bs])
                if key in self._prefix_cache:
                    self._prefix_cache.move_to_end(key)
                    continue
                blk = q.block_table[i]
                self.pool.allocator.incref(blk)
                self._prefix_cache[key] = blk

    def _evict_prefix(self, target_free: int) -> None:
        """Drop LRU cache entries until the allocator has target_free
        blocks (or the cache is empty). Freeing only drops the cache's
        own refcount — blocks shared with live sequences stay alive."""
        while (self._prefix_cache
               and self.pool.allocator.num_free < target_free):
            _, blk = self._prefix_cache.popitem(last=False)
            self.pool.allocator.free(blk)
            self._prefix_evicts += 1

    def clear_prefix_cache(self) -> None:
        if self._prefix_cache:
            self._evict_prefix(self.pool.num_blocks + 1)

    def _prefill_flat(self, seqs: List[Sequence]) -> torch.Tensor:
        """Full varlen prefill from scratch (no cached context)."""
        s = self.spec
        device = self.device
        lens = [len(q.prompt_ids) for q in seqs]
        total = sum(lens)

        # flat varlen layout
        input_ids = torch.tensor([t for q in seqs for t in q.prompt_ids],
                                 dtype=torch.long, device=device)
        positions = torch.tensor([p for L in lens for p in range(L)],
                                 dtype=torch.long, device=device)
        slot_mapping = []
        for q in seqs:
            L = len(q.prompt_ids)
            nb = KVCachePool.blocks_for(L, self.pool.block_size)
            q.block_table = [self.pool.allocator.alloc() for _ in range(nb)]
            for p in range(L):
                blk = q.block_table[p // self.pool.block_size]
                slot_mapping.append(blk * self.pool.block_size
                                    + p % self.pool.block_size)
            q.context_len = L
        slot_mapping = torch.tensor(slot_mapping, dtype=torch.long, device=device)

        x = self.model.model.embed_tokens(input_ids)
        for li, layer in enumerate(self.model.model.layers):
            h = layer.input_layernorm(x)
            q = self._proj(layer.self_attn.q_proj, h).view(total, s.num_heads, s.head_dim)
            k = self._proj(layer.self_attn.k_proj, h).view(total, s.num_kv_heads, s.head_dim)
            v = self._proj(layer.self_attn.v_proj, h).view(total, s.num_kv_heads, s.head_dim)
            if s.qk_norm:
                q, k = self._qk_norm(layer.self_attn, q, k)
            q, k = OF.apply_rope_inplace(q, k, positions, self._inv_freq,
                                         s.rope_theta)
            OF.kv_cache_scatter(k, v, self.pool.key[li], self.pool.value[li],
                                slot_mapping)
            o = self._prefill_attention(q, k, v, lens)
            x = x + self._proj(layer.self_attn.o_proj, o.reshape(total, s.q_size))
            h2 = layer.post_attention_layernorm(x)
            g = self._proj(layer.mlp.gate_proj, h2)
            u = self._proj(layer.mlp.up_proj, h2)
            x = x + self._proj(layer.mlp.down_proj, OF.silu_mul(g, u))
        x = self.model.model.norm(x)

        last_idx = torch.tensor([sum(lens[:i + 1]) - 1 for i in range(len(lens))],
                                dtype=torch.long, device=device)
        return self.model.logits(x[last_idx])

    def _qk_norm(self, at, q, k):
        """Qwen3 per-head q/k RMSNorm over head_dim, applied before RoPE
        (same op as HF Qwen3Attention.q_norm/k_norm)."""
        D = self.spec.head_dim
        q = OF.rmsnorm(q.reshape(-1, D), at.q_norm.weight,
                       self.spec.rms_norm_eps).view_as(q)
        k = OF.rmsnorm(k.reshape(-1, D), at.k_norm.weight,
                       self.spec.rms_norm_eps).view_as(k)
        return q, k

    def _prefill_attention(self, q, k, v, lens: List[int]) -> torch.Tensor:
        """Causal attention over concatenated prompts. GPU default: the
        first-party MFMA flash kernel (ops/csrc/attention.hip) on a
        per-prompt padded batch — measured 10x the varlen scalar-scoring
        kernel (1.8 vs 19.6 s for 160 ~400-token prompts) and no length
        cap. DISTRL_PREFILL_VARLEN=1 selects the legacy varlen kernel
        (<=2200 tokens) for comparison. CPU: reference varlen."""
        if (q.is_cuda and max(lens) <= 2200
                and os.environ.get("DISTRL_PREFILL_VARLEN") == "1"):
            from ..ops.build import get_extension
            ext = get_extension()
            if ext is None:
                raise RuntimeError("prefill requires the gfx950 extension")
            tq0, trows, tkv0 = [], [], []
            start = 0
            for L in lens:
                for r0 in range(0, L, 16):
                    tq0.append(start + r0)
                    trows.append(min(16, L - r0))
                    tkv0.append(start)
                start += L
            dev = q.device
            return ext.prefill_attention(
                q.contiguous(), k.contiguous(), v.contiguous(),
                torch.tensor(tq0, dtype=torch.int32, device=dev),
                torch.tensor(trows, dtype=torch.int32, device=dev),
                torch.tensor(tkv0, dtype=torch.int32, device=dev),
                max(lens), self.scale)
        if q.is_cuda:
            # prompts beyond the varlen kernel's score-tile cap: batch-pad
            # each prompt into its own row and run the first-party flash
            # kernel (ops/csrc/attention.hip) — native GQA, causal, no
            # length cap, no aotriton/SDPA (docs/ROADMAP.md #9)
            from ..ops.build import get_extension
            ext = get_extension()
            B, Lmax = len(lens), max(lens)
            H, KV, D = self.spec.num_heads, self.spec.num_kv_heads, self.spec.head_dim
            qp = q.new_zeros(B, Lmax, H, D)
            kp = k.new_zeros(B, Lmax, KV, D)
            vp = v.new_zeros(B, Lmax, KV, D)
            start = 0
            for i, L in enumerate(lens):
                qp[i, :L] = q[start:start + L]
                kp[i, :L] = k[start:start + L]
                vp[i, :L] = v[start:start + L]
                start += L
            o, _ = ext.flash_attn_fwd(
                qp.transpose(1, 2).contiguous(), kp.transpose(1, 2).contiguous(),
                vp.transpose(1, 2).contiguous(), self.scale)
            o = o.transpose(1, 2)
            # trailing pad rows never influence real rows under causal
            return torch.cat([o[i, :L] for i, L in enumerate(lens)], 0)
        return R.varlen_prefill_attention(q, k, v, lens, self.scale)

    # ------------------------------------------------------------ decode

    @torch.no_grad()
    def _decode_step(self, seqs: List[Sequence]) -> torch.Tensor:
        """One decode iteration for all running sequences: forward their
        latest sampled token, write its KV, return next-token logits."""
        s = self.spec
        device = self.device
        N = len(seqs)
        bs = self.pool.block_size

        tokens, positions, slots, ctx_lens = [], [], [], []
        max_nb = 0
        for q in seqs:
            pos = q.total_len - 1
            tokens.append(q.output_ids[-1] if q.output_ids else q.prompt_ids[-1])
            positions.append(pos)
            blk_idx = pos // bs
            if blk_idx == len(q.block_table):
                q.block_table.append(self.pool.allocator.alloc())
            slots.append(q.block_table[blk_idx] * bs + pos % bs)
            q.context_len = pos + 1
            ctx_lens.append(q.context_len)
            max_nb = max(max_nb, len(q.block_table))

        input_ids = torch.tensor(tokens, dtype=torch.long, device=device)
        positions_t = torch.tensor(positions, dtype=torch.long, device=device)
        slot_mapping = torch.tensor(slots, dtype=torch.long, device=device)
        block_tables = torch.zeros(N, max_nb, dtype=torch.int32, device=device)
        for i, q in enumerate(seqs):
            block_tables[i, :len(q.block_table)] = torch.tensor(
                q.block_table, dtype=torch.int32)
        context_lens = torch.tensor(ctx_lens, dtype=torch.int32, device=device)
        return self._decode_forward(input_ids, positions_t, slot_mapping,
                                    block_tables, context_lens)

    def _make_session(self, running, sp, eos_token_id):
        """DecodeSession per wave, with cached state buffers + captured
        hipGraph keyed on (padded N, params) reused across waves — saves
        the ~0.2-0.4 s/round re-capture cost (docs/ROADMAP.md #4).
        GPU-validated round 2: cached-vs-fresh greedy equality
        (tests/test_engine_gpu.py::test_session_cache_vs_fresh_identical)
        and identical fuzz behavior over 15 randomized trials.
        DISTRL_GRAPH_CACHE=0 reverts to per-wave sessions."""
        from .decode_session import DecodeSession, SessionCache
        if os.environ.get("DISTRL_GRAPH_CACHE", "1") == "1":
            if self._session_cache is None:
                self._session_cache = SessionCache(self)
            try:
                return self._session_cache.acquire(running, sp, eos_token_id)
            except MemoryError:
                pass  # pool too tight for the padding scratch block
        return DecodeSession(self, running, sp, eos_token_id)

    def _decode_forward(self, input_ids: torch.Tensor, positions: torch.Tensor,
                        slot_mapping: torch.Tensor, block_tables: torch.Tensor,
                        context_lens: torch.Tensor) -> torch.Tensor:
        """Single-token batched decode forward over device-side state —
        the hipGraph-capturable step body (all inputs are device tensors).

        GPU path: merged-LoRA fused weights (one qkv GEMM, one gate|up
        GEMM per layer), fused rope+KV-scatter, strided-q paged attention,
        fused residual-add+RMSNorm and packed SiLU*mul — 9 kernels/layer.
        """
        if self.device.type == "cuda":
            return self._decode_forward_fused(input_ids, positions,
                                              slot_mapping, block_tables,
                                              context_lens)
        s = self.spec
        N = input_ids.shape[0]
        x = self.model.model.embed_tokens(input_ids)
        for li, layer in enumerate(self.model.model.layers):
            h = layer.input_layernorm(x)
            q = self._proj(layer.self_attn.q_proj, h).view(N, s.num_heads, s.head_dim)
            k = self._proj(layer.self_attn.k_proj, h).view(N, s.num_kv_heads, s.head_dim)
            v = self._proj(layer.self_attn.v_proj, h).view(N, s.num_kv_heads, s.head_dim)
            if s.qk_norm:
                q, k = self._qk_norm(layer.self_attn, q, k)
            q, k = OF.apply_rope_inplace(q, k, positions, self._inv_freq,
                                         s.rope_theta)
            OF.kv_cache_scatter(k, v, self.pool.key[li], self.pool.value[li],
                                slot_mapping)
            o = OF.paged_attention_decode(q, self.pool.key[li], self.pool.value[li],
                                          block_tables, context_lens, self.scale)
            x = x + self._proj(layer.self_attn.o_proj, o.reshape(N, s.q_size))
            h2 = layer.post_attention_layernorm(x)
            g = self._proj(layer.mlp.gate_proj, h2)
            u = self._proj(layer.mlp.up_proj, h2)
            x = x + self._proj(layer.mlp.down_proj, OF.silu_mul(g, u))
        x = self.model.model.norm(x)
        return self.model.logits(x)

    def _decode_forward_fused(self, input_ids, positions, slot_mapping,
                              block_tables, context_lens):
        from ..ops.build import get_extension
        ext = get_extension()
        if ext is None:
            raise RuntimeError("fused decode requires the gfx950 extension")
        s = self.spec
        m = self.model
        qs, kvs = s.q_size, s.kv_size
        eps = s.rms_norm_eps
        lws = self.fused.layers
        nf4 = self.fused.nf4
        r = self.fused.lora_r
        pos32 = positions.to(torch.int32)
        N_batch = input_ids.shape[0]

        # per-step LoRA-u pool (zeroed ONCE per step; the split-K lora_u
        # kernels accumulate into per-(layer, site) slices)
        hybrid_down = (not nf4) and getattr(self.fused, "hybrid_down", False)
        u_pool = None
        site_off = {}
        if (nf4 or hybrid_down) and r > 0:
            off = 0
            for site in (("qkv", "o", "gateup", "down") if nf4
                         else ("down",)):
                rp = getattr(lws[0], f"{site}_r")
                site_off[site] = (off, rp)
                off += rp
            # the pool MUST keep one stable storage for the engine's
            # lifetime: decode sessions capture hipGraphs that reference
            # it, and waves of different sizes share those graphs across
            # generation rounds. A per-wave reallocation left earlier
            # waves' cached graphs pointing at freed memory (garbage u ->
            # NaN logits -> the sampler degenerates to token 0; surfaced
            # at batch 30 x 16 where pool pressure forces multiple wave
            # sizes per round). Allocate once at max_num_seqs and slice.
            if getattr(self, "_u_pool_off", None) != off:
                self._u_pool = torch.zeros(
                    len(lws), self.cfg.max_num_seqs, off,
                    dtype=torch.float32, device=input_ids.device)
                self._u_pool_off = off
            u_pool = self._u_pool[:, :N_batch, :]
            u_pool.zero_()

        def proj(x, lw, li, site, N_out, K_in):
            """One fused projection: nf4 GEMM + LoRA (or merged bf16;
            the down site runs the fused nf4 kernel even in merged mode —
            it beats hipBLASLt on that deep-K shape)."""
            if not nf4 and not (hybrid_down and site == "down"):
                if site == "qkv":
                    return F.linear(x, lw.qkv_w, lw.qkv_b)
                return F.linear(x, getattr(lw, f"{site}_w"))
            bias = lw.qkv_b if site == "qkv" else None
            if r > 0:
                o0, rt = site_off[site]
                u = u_pool[li, :, o0:o0 + rt]
                ksplit = max(1, min(8, 512 // max(1, ((N_batch + 15) // 16)
                                                 * (rt // 16))))
                ext.lora_u(x, getattr(lw, f"{site}_afrag"), u, rt, ksplit)
                return ext.nf4_gemm(x, getattr(lw, f"{site}_w4"),
                                    getattr(lw, f"{site}_amax"), bias, u,
                                    getattr(lw, f"{site}_bfrag"),
                                    N_out, K_in, rt)
            return ext.nf4_gemm(x, getattr(lw, f"{site}_w4"),
                                getattr(lw, f"{site}_amax"), bias, None,
                                None, N_out, K_in, 0)

        def qk_norm_packed(qkv, li):
            """Qwen3 per-head q/k RMSNorm applied in place on the packed
            qkv GEMM output before the fused rope+scatter. Plain torch
            ops (graph-capturable); fp32 internals match HF Qwen3. The
            norm weights are engine-lifetime parameter storages, so
            captured graphs stay valid across weight refreshes."""
            at = m.model.layers[li].self_attn
            D = s.head_dim
            for sl, w in ((qkv[:, :qs], at.q_norm.weight),
                          (qkv[:, qs:qs + kvs], at.k_norm.weight)):
                hv = sl.unflatten(1, (-1, D)).float()
                inv = hv.pow(2).mean(-1, keepdim=True).add(eps).rsqrt()
                sl.copy_((hv * inv * w.float()).to(qkv.dtype).flatten(1))

        res = m.model.embed_tokens(input_ids)
        h = ext.rmsnorm_fwd(res, lws[0].in_norm, eps)
        for li, lw in enumerate(lws):
            qkv = proj(h, lw, li, "qkv", qs + 2 * kvs, s.hidden_size)
            if s.qk_norm:
                qk_norm_packed(qkv, li)
            ext.rope_scatter_qkv(qkv, pos32, slot_mapping, self._inv_freq,
                                 self.pool.key[li], self.pool.value[li],
                                 s.num_heads, s.num_kv_heads, s.head_dim)
            attn = ext.paged_attention_decode_strided(
                qkv, s.num_heads, s.head_dim, qs + 2 * kvs,
                self.pool.key[li], self.pool.value[li], block_tables,
                context_lens, self.scale)
            o = proj(attn.view(-1, qs), lw, li, "o", s.hidden_size, qs)
            h, res = ext.add_rmsnorm_fwd(res, o, lw.post_norm, eps)
            gu = proj(h, lw, li, "gateup", 2 * s.intermediate_size,
                      s.hidden_size)
            act = ext.silu_mul_packed(gu)
            d = proj(act, lw, li, "down", s.hidden_size, s.intermediate_size)
            next_w = (lws[li + 1].in_norm if li + 1 < len(lws)
                      else m.model.norm.weight)
            h, res = ext.add_rmsnorm_fwd(res, d, next_w, eps)
        return m.logits(h)

    # ----------------------------------------------------------- forking

    def _fork(self, parent: Sequence, n: int, first_tokens: List[int]) -> List[Sequence]:
        """Create n candidate sequences sharing the parent's full prompt
        blocks; a partial tail block is copied per candidate."""
        bs = self.pool.block_size
        L = len(parent.prompt_ids)
        full = L // bs  # number of completely-filled blocks
        children = []
        for i in range(n):
            self._seq_counter += 1
            child = Sequence(self._seq_counter, parent.prompt_ids,
                             parent.parent_prompt)
            child.cand_index = i
            table = []
            for b in parent.block_table[:full]:
                self.pool.allocator.incref(b)
                table.append(b)
            if full < len(parent.block_table):
                nb = self.pool.allocator.alloc()
                self.pool.copy_block(parent.block_table[full], nb)
                table.append(nb)
            child.block_table = table
            child.context_len = L
            child.output_ids = [first_tokens[i]]
            children.append(child)
        # parent's own references are released (children hold their own)
        for b in parent.block_table:
            self.pool.allocator.free(b)
        parent.block_table = []
        return children

    def _finish(self, seq: Sequence) -> None:
        seq.finished = True
        for b in seq.block_table:
            self.pool.allocator.free(b)
        seq.block_table = []

    # ---------------------------------------------------------- generate

    @torch.no_grad()
    def generate(self, prompts: List[List[int]], sp: SamplingParams,
                 eos_token_id: Optional[int] = None,
                 prefill_token_budget: int = 8192,
                 stream_cb=None,
                 token_limits: Optional[List[List[int]]] = None,
                 cancel_check=None
                 ) -> List[List[List[int]]]:
        """Generate sp.n completions per prompt.

        prompts: token-id lists. Returns per prompt a list of n output
        token-id lists (EOS included when emitted, like vLLM's
        ``o.token_ids``).

        stream_cb: optional ``f(prompt_index, cand_index, new_token_ids)``
        called as tokens become known — per token on the eager path, per
        decode chunk on the session path (serving SSE streaming).

        token_limits: optional per-prompt lists of per-candidate output
        caps (each <= sp.max_tokens); finished lanes retire in-wave and
        free their decode slots (the serving batcher uses this to merge
        requests with different max_tokens; the EOS-realistic bench mode
        draws them from an exponential via sp.geom_len_mean).

        cancel_check: optional ``f(prompt_index) -> bool`` polled between
        decode chunks (request abort, e.g. an SSE client disconnecting —
        vLLM's ``abort_request`` analogue). A cancelled prompt's
        candidates finish immediately with their partial outputs (or
        ``[]`` if never admitted) and their KV blocks are freed; abort
        latency is bounded by one decode chunk (16 steps).
        """
        was_training = self.model.training
        self.model.eval()
        if self.fused is not None:
            # fold the current LoRA into the merged decode weights
            # (once per generation round, never inside the decode loop)
            self.fused.refresh()
        try:
            return self._generate_inner(prompts, sp, eos_token_id,
                                        prefill_token_budget, stream_cb,
                                        token_limits, cancel_check)
        except Exception:
            # a failure mid-generation strands this call's in-flight
            # sequences' KV blocks; generate calls are serialized, so no
            # other sequence is live — reset the pool instead of leaking
            # (long-lived serving processes would otherwise exhaust it)
            self.pool.allocator.reset()
            if self._prefix_cache:
                # the reset dropped every refcount; cached block ids are
                # stale — start the cache over rather than dangle
                self._prefix_cache.clear()
            raise
        finally:
            if was_training:
                self.model.train()

    def _generate_inner(self, prompts, sp, eos_token_id, prefill_token_budget,
                        stream_cb=None, token_limits=None, cancel_check=None):
        # token_limits: optional per-prompt lists of per-candidate output
        # caps (each <= sp.max_tokens) — the EOS-realistic bench mode and
        # per-request limits use these; admission still reserves the
        # worst case from sp.max_tokens.
        if token_limits is not None:
            if len(token_limits) != len(prompts):
                raise ValueError("token_limits must have one list per prompt")
            for per in token_limits:
                if len(per) != sp.n or any(
                        not (1 <= int(v) <= sp.max_tokens) for v in per):
                    raise ValueError(
                        "each token_limits entry needs sp.n caps in "
                        f"[1, {sp.max_tokens}]")
        if sp.n > self.cfg.max_num_seqs:
            # the n-candidate fan-out of one prompt is admitted atomically,
            # so it can never fit — fail with the real reason instead of
            # the admission loop's pool-exhaustion error
            raise ValueError(
                f"SamplingParams.n={sp.n} exceeds max_num_seqs="
                f"{self.cfg.max_num_seqs}; raise EngineConfig.max_num_seqs")
        if sp.seed is not None:
            # per-request determinism (vLLM SamplingParams.seed analogue):
            # reseed this call's sampling stream so identical (prompts,
            # params, seed) calls reproduce regardless of engine history
            self.generator.manual_seed(int(sp.seed))
        bs = self.pool.block_size
        max_total = self.cfg.max_seq_length
        # results indexed by (prompt, candidate): with EOS / per-seq
        # limits and in-wave retirement, candidates finish out of order —
        # the contract (like vLLM's) is CANDIDATE order
        results: List[List[List[int]]] = [[None] * sp.n for _ in prompts]

        waiting = list(range(len(prompts)))
        running: List[Sequence] = []

        def blocks_needed(pi: int) -> int:
            L = min(len(prompts[pi]), max_total - 1)
            worst = min(L + sp.max_tokens, max_total)
            return (KVCachePool.blocks_for(L, bs)
                    + sp.n * (KVCachePool.blocks_for(worst, bs)
                              - L // bs))

        def future_need() -> int:
            """Blocks the already-running sequences may still allocate
            (worst case) — reserved so decode never exhausts the pool."""
            need = 0
            for q in running:
                worst = min(len(q.prompt_ids) + sp.max_tokens, max_total)
                need += KVCachePool.blocks_for(worst, bs) - len(q.block_table)
            return need

        def try_admit():
            """Prefill + fork as many waiting prompts as memory allows
            (continuous batching admission)."""
            if cancel_check is not None:
                # purge cancelled prompts that never started: their
                # candidates resolve to empty outputs (a prompt is
                # admitted atomically, so all its slots are still None)
                for pi in [p for p in waiting if cancel_check(p)]:
                    waiting.remove(pi)
                    results[pi] = [[] for _ in range(sp.n)]
            batch: List[Sequence] = []
            batch_tokens = 0
            reserved = future_need()
            while waiting:
                pi = waiting[0]
                need = blocks_needed(pi)
                L = min(len(prompts[pi]), max_total - 1)
                if self._prefix_cache:
                    # reclaim LRU-cached prefix blocks before refusing
                    self._evict_prefix(need + reserved + len(batch))
                if need > self.pool.allocator.num_free - reserved - len(batch):
                    break
                reserved += need
                if batch and batch_tokens + L > prefill_token_budget:
                    break
                if len(running) + (len(batch) + 1) * sp.n > self.cfg.max_num_seqs:
                    break
                waiting.pop(0)
                self._seq_counter += 1
                seq = Sequence(self._seq_counter, prompts[pi][:L], pi)
                batch.append(seq)
                batch_tokens += L
            if not batch:
                return
            with trace_range("engine/prefill"):
                logits = self._prefill_batch(batch)
            for i, parent in enumerate(batch):
                lg = logits[i:i + 1].expand(sp.n, -1).contiguous()
                first = OF.sample_tokens(lg, sp.temperature, sp.top_p, sp.top_k,
                                         generator=self.generator)
                children = self._fork(parent, sp.n, first.tolist())
                if token_limits is not None:
                    for ci, c in enumerate(children):
                        c.max_tokens = token_limits[parent.parent_prompt][ci]
                elif sp.geom_len_mean:
                    # EOS-realistic mode: per-candidate exponential caps
                    uu = torch.rand(len(children), generator=self.generator,
                                    device=self.device)
                    lims = (1.0 - float(sp.geom_len_mean)
                            * uu.clamp_min(1e-9).log()).long() \
                        .clamp(1, sp.max_tokens)
                    for c, lim in zip(children, lims.tolist()):
                        c.max_tokens = int(lim)
                for c in children:
                    if stream_cb is not None:
                        stream_cb(c.parent_prompt, c.cand_index,
                                  [c.output_ids[0]])
                    if ((eos_token_id is not None
                         and c.output_ids[-1] == eos_token_id)
                            or len(c.output_ids) >= (c.max_tokens
                                                     or sp.max_tokens)
                            or c.total_len >= max_total):
                        # still need its KV? No: sequence is done.
                        results[c.parent_prompt][c.cand_index] = c.output_ids
                        self._finish(c)
                    else:
                        running.append(c)

        try_admit()

        force_session = os.environ.get("DISTRL_FORCE_SESSION") == "1"
        if ((self.device.type == "cuda" and not self.cfg.enforce_eager)
                or force_session):
            # wave-based decode sessions (device state + hipGraph replay;
            # DISTRL_FORCE_SESSION=1 runs the same state machine on CPU
            # so CI covers the graph step body without a GPU).
            # In-wave continuous batching: a wave exits early once
            # `retire_unit` lanes finish; finished lanes are retired,
            # waiting prompts admitted, and the survivors re-waved (the
            # session cache keeps the per-padded-size graphs, so a
            # re-wave costs a reset, not a re-capture). Reference parity:
            # vLLM's continuous batching inside fast_generate
            # (distributed_actor.py:147-172).
            retire_unit = int(os.environ.get("DISTRL_RETIRE_UNIT", "16"))
            while running or waiting:
                if not running:
                    try_admit()
                    if not running:
                        if waiting:
                            raise MemoryError(
                                "KV pool too small to admit any waiting prompt")
                        break
                session = self._make_session(running, sp, eos_token_id)
                retire_at = retire_unit if retire_unit > 0 else None
                stop = None
                if cancel_check is not None:
                    wave = running
                    stop = lambda: any(cancel_check(q.parent_prompt)
                                       for q in wave)
                try:
                    with trace_range(f"engine/decode_wave[{len(running)}]"):
                        new_toks, fin = session.run(stream_cb=stream_cb,
                                                    retire_at=retire_at,
                                                    stop_check=stop)
                finally:
                    if self._session_cache is not None:
                        from .decode_session import CachedDecodeSession
                        if isinstance(session, CachedDecodeSession):
                            self._session_cache.release(session)
                still = []
                for q, toks, f in zip(running, new_toks, fin):
                    q.output_ids.extend(toks)
                    if cancel_check is not None and \
                            cancel_check(q.parent_prompt):
                        f = True  # aborted: partial output, free the KV
                    if f:
                        results[q.parent_prompt][q.cand_index] = q.output_ids
                        self._finish(q)
                    else:
                        still.append(q)
                running = still
                if waiting:
                    try_admit()  # mid-flight admission into the next wave
            return results

        while running or waiting:
            if not running:
                try_admit()
                if not running:
                    if waiting:
                        raise MemoryError(
                            "KV pool too small to admit any waiting prompt")
                    break
            logits = self._decode_step(running)
            next_tokens = OF.sample_tokens(logits, sp.temperature, sp.top_p,
                                           sp.top_k, generator=self.generator)
            next_list = next_tokens.tolist()
            still = []
            for i, q in enumerate(running):
                t = next_list[i]
                q.output_ids.append(t)
                if stream_cb is not None:
                    stream_cb(q.parent_prompt, q.cand_index, [t])
                done = ((eos_token_id is not None and t == eos_token_id)
                        or len(q.output_ids) >= (q.max_tokens or sp.max_tokens)
                        or q.total_len >= max_total
                        or (cancel_check is not None
                            and cancel_check(q.parent_prompt)))
                if done:
                    results[q.parent_prompt][q.cand_index] = q.output_ids
                    self._finish(q)
                else:
                    still.append(q)
            running = still
            if waiting:
                try_admit()

        return results
