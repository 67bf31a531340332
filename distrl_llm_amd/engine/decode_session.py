"""Device-resident decode session with hipGraph-captured steps.

The eager decode loop pays per-step Python state assembly + ~700 kernel
launches per step on a 28-layer model. A DecodeSession instead keeps ALL
per-sequence state (tokens, positions, context lengths, block tables,
finished mask, output buffer, sampler seed/step counters) in device
tensors, pre-allocates each sequence's worst-case KV blocks up front (the
admission control already reserved them), and expresses one decode step as
a fixed tensor program over those buffers — embed -> 28 layers (HIP
rmsnorm / rope / kv-scatter / paged-attention + hipBLASLt projections) ->
LM head -> fused sampler -> in-graph state advance. That program is
captured once into a hipGraph and replayed per token (SURVEY.md §2.4-A:
"hipGraph-captured decode step" north star); the host loop is just
graph.replay() with one finished-mask sync per chunk of steps.

Sampling randomness inside the graph: the fused sampler hashes
(per-seq seed, device step counter), and the counter increments in-graph,
so every replay draws fresh randomness with zero host work.

Frozen (finished) sequences stop advancing their position and re-forward
their last token idempotently; their recorded garbage steps are dropped at
extraction (per-seq output length = final_position - prompt_len + 1).
"""

from __future__ import annotations

from typing import List, Optional

import torch

from ..config import SamplingParams
from .kvcache import KVCachePool, Sequence


class DecodeSession:
    def __init__(self, engine, seqs: List[Sequence], sp: SamplingParams,
                 eos_token_id: Optional[int], use_graph: bool = True):
        self.engine = engine
        self.seqs = seqs
        self.sp = sp
        self.eos = eos_token_id
        dev = engine.device
        bs = engine.pool.block_size
        N = len(seqs)
        max_total = engine.cfg.max_seq_length

        # sequences may be RESUMED mid-generation (in-wave retirement re-
        # waves survivors): the session starts each lane at its CURRENT
        # last-token position, not at the prompt boundary
        pos0 = [q.total_len - 1 for q in seqs]
        limits = [min(len(q.prompt_ids) + (q.max_tokens or sp.max_tokens),
                      max_total) - 1 for q in seqs]

        # pre-allocate every block each sequence can ever need
        for q, lim in zip(seqs, limits):
            need = KVCachePool.blocks_for(lim + 1, bs)
            while len(q.block_table) < need:
                q.block_table.append(engine.pool.allocator.alloc())
        max_nb = max(len(q.block_table) for q in seqs)

        bt = torch.zeros(N, max_nb, dtype=torch.int32)
        for i, q in enumerate(seqs):
            bt[i, :len(q.block_table)] = torch.tensor(q.block_table,
                                                      dtype=torch.int32)
        self.block_tables = bt.to(dev, non_blocking=True)
        self.tokens = torch.tensor([q.output_ids[-1] for q in seqs],
                                   dtype=torch.long, device=dev)
        self.positions = torch.tensor(pos0, dtype=torch.long, device=dev)
        self.pos0 = pos0
        self.ctx_lens = (self.positions + 1).to(torch.int32)
        self.limit_pos = torch.tensor(limits, dtype=torch.long, device=dev)
        self.finished = torch.zeros(N, dtype=torch.bool, device=dev)
        self.step_idx = torch.zeros(1, dtype=torch.long, device=dev)
        self.seeds = torch.randint(0, 2**31 - 1, (N,), dtype=torch.int64,
                                   device=dev, generator=engine.generator)
        self.max_steps = max(lim - p0 for lim, p0 in zip(limits, pos0))
        # >= 2 rows: _capture() warms up with TWO _step() calls before the
        # state restore, so step_idx reaches 1 — a 1-row buffer would make
        # the warmup's index_copy_ write out of bounds (an async device
        # fault that surfaces as an HSA exception a trial later)
        self.out_buf = torch.zeros(max(self.max_steps, 2), N,
                                   dtype=torch.long, device=dev)
        self.graph = None
        self.use_graph = use_graph and dev.type == "cuda"

    # ------------------------------------------------------------- step

    def _step(self):
        e = self.engine
        bs = e.pool.block_size
        pos = self.positions
        blk = torch.div(pos, bs, rounding_mode="floor")
        slot = (self.block_tables.gather(1, blk.unsqueeze(1)).squeeze(1).long()
                * bs + pos % bs)
        logits = e._decode_forward(self.tokens, pos, slot, self.block_tables,
                                   self.ctx_lens)
        if self.sp.temperature > 0.0:
            if logits.is_cuda:
                from ..ops.build import get_extension
                ext = get_extension()
                # two-stage sampler fills the chip at decode batch
                # sizes (160: 306 -> 242 us; 64: 299 -> 135 us); the
                # single-kernel version wins again once B alone covers
                # the CUs (512: 576 vs 626 us)
                fn = (ext.sample_tokens2 if logits.shape[0] <= 256
                      else ext.sample_tokens)
                sampled = fn(logits, float(self.sp.temperature),
                             float(self.sp.top_p),
                             int(self.sp.top_k), self.seeds,
                             self.step_idx)
            else:
                # CPU session (DISTRL_FORCE_SESSION=1 CI coverage of the
                # state machine): reference sampler on the engine stream
                from ..ops import functional as OF
                sampled = OF.sample_tokens(logits,
                                           float(self.sp.temperature),
                                           float(self.sp.top_p),
                                           int(self.sp.top_k),
                                           generator=e.generator)
        else:
            sampled = logits.argmax(-1)
        self.out_buf.index_copy_(0, self.step_idx, sampled.unsqueeze(0))

        was_finished = self.finished
        new_fin = was_finished | (self.positions + 1 >= self.limit_pos)
        if self.eos is not None:
            new_fin = new_fin | (~was_finished & sampled.eq(self.eos))
        adv = (~was_finished).long()
        self.tokens.copy_(torch.where(was_finished, self.tokens, sampled))
        self.positions.add_(adv)
        self.ctx_lens.add_(adv.to(torch.int32))
        self.finished.copy_(new_fin)
        self.step_idx.add_(1)

    # ---------------------------------------------------------- capture

    def _capture(self):
        state = [self.tokens, self.positions, self.ctx_lens, self.finished,
                 self.step_idx]
        saved = [t.clone() for t in state]
        torch.cuda.synchronize()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            self._step()
            self._step()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        for t, sv in zip(state, saved):
            t.copy_(sv)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._step()
        self.graph = g

    # -------------------------------------------------------------- run

    def run(self, chunk: int = 16, stream_cb=None, retire_at=None,
            stop_check=None):
        """Decode until every lane finishes, or — when ``retire_at`` is
        set — until at least that many lanes have finished (in-wave
        retirement: the engine then retires them, admits waiting prompts
        and re-waves the survivors; reference parity: vLLM continuous
        batching inside fast_generate, distributed_actor.py:147-172).

        stop_check: optional zero-arg callable polled once per chunk —
        returning True exits the wave early (request cancellation: the
        engine then retires the cancelled lanes with partial output,
        bounding abort latency to one chunk of steps).

        Returns (new_tokens_per_seq, finished_per_seq): only the tokens
        generated by THIS session (the caller extends q.output_ids)."""
        reported = [0] * len(self.seqs)  # out_buf rows already streamed
        if self.max_steps > 0:
            if self.use_graph and self.graph is None:
                try:
                    self._capture()
                except Exception as err:  # capture unsupported -> eager
                    import sys
                    print(f"[engine] hipGraph capture failed ({err}); "
                          f"eager decode", file=sys.stderr)
                    self.graph = None
            steps = 0
            while steps < self.max_steps:
                n = min(chunk, self.max_steps - steps)
                if self.graph is not None:
                    for _ in range(n):
                        self.graph.replay()
                else:
                    for _ in range(n):
                        self._step()
                steps += n
                if stream_cb is not None:
                    # chunk-granular streaming: emit newly valid rows
                    pos = self.positions.cpu().tolist()
                    buf = self.out_buf[:steps].cpu()
                    for i, q in enumerate(self.seqs):
                        n_new = pos[i] - self.pos0[i]
                        if n_new > reported[i]:
                            stream_cb(q.parent_prompt, q.cand_index,
                                      buf[reported[i]:n_new, i].tolist())
                            reported[i] = n_new
                n_fin = int(self.finished.sum())
                if n_fin >= self.finished.numel():  # incl. padded dummies
                    break
                if retire_at is not None and n_fin >= retire_at:
                    break
                if stop_check is not None and stop_check():
                    break

        # ---- extraction: this session's new tokens + finished flags ----
        final_pos = self.positions.cpu().tolist()
        fin = self.finished.cpu().tolist()
        out = self.out_buf.cpu()
        new_tokens = []
        for i, q in enumerate(self.seqs):
            n_new = final_pos[i] - self.pos0[i]
            new_tokens.append(out[:n_new, i].tolist())
        return new_tokens, fin[:len(self.seqs)]


# --------------------------------------------------------- session cache

class CachedDecodeSession(DecodeSession):
    """A DecodeSession whose device buffers (and captured hipGraph) are
    owned by a SessionCache and REUSED across generation waves
    (docs/ROADMAP.md #4: capture cost ~0.2-0.4 s/round otherwise).

    Shapes are fixed at worst case for the cache key — N padded up, block
    tables at blocks_for(max_seq_length), out_buf at sp.max_tokens — so a
    graph captured once stays valid; ``reset`` refills the same storages
    for each new wave. Lanes beyond the real batch are born finished and
    point at a per-wave scratch KV block (frozen lanes re-forward
    idempotently, so their writes land in the scratch block and their
    outputs are never extracted).

    Gated off by default (DISTRL_GRAPH_CACHE=1): the padding/reset state
    machine is CPU-tested via DISTRL_FORCE_SESSION; flipping the default
    needs a GPU validation pass (round-2).
    """

    def __init__(self, engine, n_pad: int, sp: SamplingParams,
                 eos_token_id: Optional[int]):
        # deliberately NOT calling DecodeSession.__init__: buffers are
        # allocated once at worst-case shapes, then refilled by reset()
        self.engine = engine
        self.sp = sp
        self.eos = eos_token_id
        dev = engine.device
        bs = engine.pool.block_size
        self.n_pad = n_pad
        max_total = engine.cfg.max_seq_length
        max_nb = KVCachePool.blocks_for(max_total, bs)
        self.block_tables = torch.zeros(n_pad, max_nb, dtype=torch.int32,
                                        device=dev)
        self.tokens = torch.zeros(n_pad, dtype=torch.long, device=dev)
        self.positions = torch.zeros(n_pad, dtype=torch.long, device=dev)
        self.ctx_lens = torch.ones(n_pad, dtype=torch.int32, device=dev)
        self.limit_pos = torch.zeros(n_pad, dtype=torch.long, device=dev)
        self.finished = torch.zeros(n_pad, dtype=torch.bool, device=dev)
        self.step_idx = torch.zeros(1, dtype=torch.long, device=dev)
        self.seeds = torch.zeros(n_pad, dtype=torch.int64, device=dev)
        # >= 2 rows for the same capture-warmup reason as the base class
        self.out_buf = torch.zeros(max(sp.max_tokens, 2), n_pad,
                                   dtype=torch.long, device=dev)
        self.graph = None
        self.use_graph = dev.type == "cuda"
        self.seqs: List[Sequence] = []
        self.pos0: List[int] = []
        self.max_steps = 0
        self._scratch_block: Optional[int] = None

    def reset(self, seqs: List[Sequence], scratch_block: Optional[int]):
        """Refill the cached buffers for a new wave (same storages, so a
        previously captured graph remains valid)."""
        engine = self.engine
        bs = engine.pool.block_size
        sp = self.sp
        N = len(seqs)
        assert N <= self.n_pad
        max_total = engine.cfg.max_seq_length
        pos0 = [q.total_len - 1 for q in seqs]
        limits = [min(len(q.prompt_ids) + (q.max_tokens or sp.max_tokens),
                      max_total) - 1 for q in seqs]
        for q, lim in zip(seqs, limits):
            need = KVCachePool.blocks_for(lim + 1, bs)
            while len(q.block_table) < need:
                q.block_table.append(engine.pool.allocator.alloc())

        self.seqs = seqs
        self.pos0 = pos0
        self.max_steps = max(lim - p0 for lim, p0 in zip(limits, pos0))
        self._scratch_block = scratch_block

        n_pad, dev = self.n_pad, engine.device
        bt = torch.zeros(n_pad, self.block_tables.shape[1], dtype=torch.int32)
        for i, q in enumerate(seqs):
            bt[i, :len(q.block_table)] = torch.tensor(q.block_table,
                                                      dtype=torch.int32)
        if N < n_pad:
            # dummy lanes: frozen from step 0, KV writes land in scratch
            sb = scratch_block if scratch_block is not None else 0
            bt[N:, 0] = sb
        self.block_tables.copy_(bt.to(dev, non_blocking=True))

        def fill(t, vals, pad):
            host = torch.full((n_pad,), pad, dtype=t.dtype)
            host[:N] = torch.tensor(vals, dtype=t.dtype)
            t.copy_(host.to(dev, non_blocking=True))

        fill(self.tokens, [q.output_ids[-1] for q in seqs], 0)
        fill(self.positions, pos0, 0)
        fill(self.ctx_lens, [p + 1 for p in pos0], 1)
        fill(self.limit_pos, limits, 0)  # dummy limit 0 -> finished
        host_fin = torch.zeros(n_pad, dtype=torch.bool)
        host_fin[N:] = True
        self.finished.copy_(host_fin.to(dev, non_blocking=True))
        self.step_idx.zero_()
        # generate on the engine's device: its generator is device-bound
        # (a CUDA generator cannot seed a CPU-side randint)
        self.seeds.copy_(torch.randint(0, 2**31 - 1, (n_pad,),
                                       dtype=torch.int64, device=dev,
                                       generator=engine.generator))
        self.out_buf.zero_()
        return self

    # run() is inherited: the base loop only captures when no graph
    # exists yet, so a graph captured on a previous wave (same storages)
    # replays directly, and the padded dummy lanes are born finished so
    # the retirement/extraction logic never sees them (finished count is
    # compared against n_pad via retire_at offsetting in SessionCache).

    def run(self, chunk: int = 16, stream_cb=None, retire_at=None,
            stop_check=None):
        # dummy lanes count as finished: shift the retirement threshold
        if retire_at is not None:
            retire_at = retire_at + (self.n_pad - len(self.seqs))
        return super().run(chunk=chunk, stream_cb=stream_cb,
                           retire_at=retire_at, stop_check=stop_check)


class SessionCache:
    """Per-engine cache of CachedDecodeSession state keyed on
    (padded batch size, sampling params, eos). One scratch KV block is
    borrowed from the pool per wave for the padding lanes and returned
    after."""

    def __init__(self, engine):
        self.engine = engine
        self._cache = {}

    @staticmethod
    def _pad(n: int) -> int:
        """Pad the wave to a multiple of 16 (min 8). Power-of-two padding
        measured a 12% END-TO-END regression on the headline bench: batch
        160 padded to 256 runs every decode GEMM/attention 60% fatter
        (gen 13.5 -> 17.0 s/round). Multiple-of-16 keeps the graph-count
        bounded (<= max_num_seqs/16 cache entries) at near-zero padding
        waste."""
        if n <= 8:
            return 8
        return (n + 15) // 16 * 16

    def acquire(self, seqs: List[Sequence], sp: SamplingParams,
                eos_token_id: Optional[int]) -> CachedDecodeSession:
        n_pad = self._pad(len(seqs))
        key = (n_pad, sp.max_tokens, sp.temperature, sp.top_p, sp.top_k,
               eos_token_id)
        sess = self._cache.get(key)
        if sess is None:
            sess = CachedDecodeSession(self.engine, n_pad, sp, eos_token_id)
            self._cache[key] = sess
        scratch = None
        if n_pad > len(seqs):
            scratch = self.engine.pool.allocator.alloc()
        try:
            return sess.reset(seqs, scratch)
        except MemoryError:
            # per-seq worst-case prealloc failed after the scratch was
            # taken — return it before falling back to an exact session
            if scratch is not None:
                self.engine.pool.allocator.free(scratch)
            raise

    def release(self, sess: CachedDecodeSession):
        if sess._scratch_block is not None:
            self.engine.pool.allocator.free(sess._scratch_block)
            sess._scratch_block = None
