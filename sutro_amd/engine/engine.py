"""LLMEngine: continuous-batching execution loop (generative + embedding).

One engine per GPU process. `step()` runs one scheduler iteration: build the
flat token batch, forward the model, sample (or pool, for embedding models),
update request state, and return per-step stats. The service layer drives it
from a worker thread/process; `bench.py` drives it directly.

With `EngineConfig.async_decode`, pure-decode steps run one-step-lagged: the
sampled-token tensor of step N feeds step N+1's input ids directly (no host
round-trip or device sync on the critical path) and the host applies step
N's tokens while N+1 executes. Rows that finish decode one extra discarded
token; FSM-guided rows and mixed prefill steps take the synchronous path.
Sync-equivalence is proven by tests (seeded rows, incl. under preemption).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np
import torch

from ..models.qwen3 import Qwen3Model
from ..models.registry import ModelSpec
from .batch import ForwardBatch, ScheduledBatch
from .config import EngineConfig
from .guided import GuidedFSM
from .kv_cache import PagedKVCache
from .request import FinishReason, Request, SamplingParams
from .sampler import Sampler
from .scheduler import Scheduler
from .tokenizer import EOS_ID, get_tokenizer


@dataclass
class StepStats:
    scheduled_tokens: int = 0
    prefill_tokens: int = 0
    output_tokens: int = 0
    finished: List[Request] = field(default_factory=list)


def _setup_tunableop(cfg: EngineConfig) -> None:
    """Enable PyTorch TunableOp (hipBLASLt algorithm selection) with the
    shipped gfx950 tuning table. Set SUTRO_AMD_TUNABLEOP_TUNE=1 to re-tune
    (slow warmup; call torch.cuda.tunable.write_file() afterwards)."""
    if not cfg.device.startswith("cuda"):
        return
    import os as _os

    if _os.environ.get("SUTRO_AMD_TUNABLEOP", "1") == "0":
        return
    try:
        import torch.cuda.tunable as tunable
    except ImportError:
        return
    path = _os.environ.get(
        "SUTRO_AMD_TUNABLEOP_FILE",
        _os.path.join(_os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))),
                      "data", "tunableop_gfx950.csv"))
    tunable.enable(True)
    tunable.set_filename(path)
    if _os.path.exists(path):
        tunable.read_file(path)
    tunable.tuning_enable(_os.environ.get("SUTRO_AMD_TUNABLEOP_TUNE", "0") == "1")


class LLMEngine:
    def __init__(self, cfg: EngineConfig, model: Optional[torch.nn.Module] = None):
        self.cfg = cfg
        self.spec: ModelSpec = cfg.spec
        self.device = cfg.device
        # tokenizer truncated to the model vocab (BPE merge-prefix property);
        # models with vocab > the shipped tokenizer get the full tokenizer and
        # the sampler masks the dead tail
        self.tokenizer = get_tokenizer(self.spec.vocab_size)
        _setup_tunableop(cfg)
        from ..parallel.tp import TPContext

        self.tp = (TPContext.from_world(cfg.tp_size) if cfg.tp_size > 1
                   else TPContext())
        if model is None:
            with torch.device(cfg.device):
                model = Qwen3Model(self.spec, cfg.dtype, cfg.max_model_len,
                                   self.tp, moe_ep=cfg.moe_ep)
            if cfg.weights_path:
                from ..models.loader import load_weights

                load_weights(model, cfg.weights_path)
            else:
                model.init_random_weights(cfg.seed)
        self.model = model.to(cfg.device).eval()

        num_blocks = cfg.num_kv_blocks
        if num_blocks is None:
            if cfg.device.startswith("cuda"):
                free, total = torch.cuda.mem_get_info()
                headroom = int((1.0 - cfg.gpu_memory_utilization) * total)
                num_blocks = cfg.derive_num_kv_blocks(max(0, free - headroom))
                # cap the block-table width the scheduler can ever need
                max_blocks_useful = (
                    (cfg.max_model_len + cfg.kv_block_size - 1) // cfg.kv_block_size
                ) * cfg.max_num_seqs
                num_blocks = min(num_blocks, max_blocks_useful)
            else:
                num_blocks = 1024
        kvh = self.spec.num_kv_heads
        if cfg.tp_size > 1:
            kvh = max(1, kvh // cfg.tp_size)
        self.kv = PagedKVCache(
            num_layers=self.spec.num_layers,
            num_blocks=num_blocks,
            num_kv_heads=kvh,
            block_size=cfg.kv_block_size,
            head_dim=self.spec.head_dim,
            dtype=cfg.kv_torch_dtype(),
            device=cfg.device,
        )
        # block 0 is reserved scratch: hipGraph padding rows read/write it
        self.scratch_block = self.kv.allocator.allocate(1)[0]
        self.scheduler = Scheduler(cfg, self.kv)
        self.sampler = Sampler(cfg.device, seed=cfg.seed,
                               vocab_limit=self.tokenizer.vocab_size)

        # size the shared MoE workspace BEFORE any hipGraph capture: big
        # transients allocated inside a capture become graph-owned per
        # bucket per layer (measured ~277 GB on qwen-3-30b-a3b, call 9)
        if cfg.device.startswith("cuda"):
            from ..models.qwen3 import Qwen3MoE

            for mod in self.model.modules():
                if isinstance(mod, Qwen3MoE):
                    mod.prealloc_workspace(max(cfg.max_tokens_per_step,
                                               cfg.max_num_seqs))
                    break  # workspace is shared across layers

        self.graph_runner = None
        if (cfg.device.startswith("cuda") and not cfg.enforce_eager
                and not self.spec.embedding):
            from .graph_runner import DecodeGraphRunner

            try:
                self.graph_runner = DecodeGraphRunner(self)
            except Exception as e:
                import warnings

                warnings.warn(f"hipGraph decode capture failed ({e}); "
                              f"running decode eagerly")
                self.graph_runner = None

        self.prefill_graph = None
        if (cfg.graph_prefill and cfg.device.startswith("cuda")
                and not cfg.enforce_eager):
            from .prefill_graph import PrefillGraphRunner

            try:
                self.prefill_graph = PrefillGraphRunner(self)
                self.prefill_graph.capture(
                    pool=self.graph_runner._pool
                    if self.graph_runner is not None else None)
            except Exception as e:
                import warnings

                warnings.warn(f"hipGraph prefill capture failed ({e}); "
                              f"running prefill eagerly")
                self.prefill_graph = None

        self._next_req_id = 0
        self._full_mask_row: Optional[torch.Tensor] = None
        # async decode: step N's sampled-token tensors, consumed at N+1
        # (reqs, tokens_t, lps_t, alloc_gens)
        self._pending_decode: Optional[tuple] = None
        self._fsms: Dict[int, GuidedFSM] = {}
        self._next_fsm_id = 0
        self._emb_sums: Dict[int, torch.Tensor] = {}
        self.embeddings: Dict[int, np.ndarray] = {}

        # cumulative stats
        self.total_prompt_tokens = 0
        self.total_output_tokens = 0

    # ---- admission ----

    def register_fsm(self, schema: dict) -> int:
        fsm = GuidedFSM.from_schema(schema, tokenizer=self.tokenizer,
                                    device=self.device)
        fsm_id = self._next_fsm_id
        self._next_fsm_id += 1
        self._fsms[fsm_id] = fsm
        return fsm_id

    def add_request(
        self,
        prompt_token_ids: List[int],
        sampling: Optional[SamplingParams] = None,
        fsm_id: Optional[int] = None,
        priority: int = 0,
        arrival_idx: int = 0,
        truncate: bool = True,
    ) -> Request:
        sampling = sampling or SamplingParams(max_tokens=self.cfg.default_max_new_tokens)
        # truncation reserves the FULL requested generation budget (capped so
        # a huge max_tokens still leaves a 16-token prompt window) — a 64-cap
        # here made near-limit rows finish early with LENGTH
        limit = max(16, self.cfg.max_model_len - max(1, sampling.max_tokens))
        if len(prompt_token_ids) > limit:
            if truncate:
                prompt_token_ids = prompt_token_ids[:limit]
            else:
                raise ValueError(
                    f"prompt of {len(prompt_token_ids)} tokens exceeds max_model_len "
                    f"{self.cfg.max_model_len} and truncate_rows is False"
                )
        req = Request(
            req_id=self._next_req_id,
            prompt_token_ids=list(prompt_token_ids),
            sampling=sampling,
            fsm_id=fsm_id,
            arrival_idx=arrival_idx,
        )
        if fsm_id is not None:
            req.fsm_state = self._fsms[fsm_id].start_state()
            req.fsm_start_state = req.fsm_state
        self._next_req_id += 1
        self.scheduler.add_request(req, priority)
        return req

    def abort_request(self, req: Request) -> None:
        self.scheduler.abort_request(req)

    def has_work(self) -> bool:
        return self.scheduler.has_work()

    # ---- the step ----

    def _build_forward_batch(self, sb: ScheduledBatch) -> ForwardBatch:
        S = len(sb.reqs)
        T = sb.total_tokens
        input_ids = np.empty(T, dtype=np.int64)
        positions = np.empty(T, dtype=np.int64)
        slots = np.empty(T, dtype=np.int64)
        qlocs = np.zeros(S + 1, dtype=np.int32)
        seq_lens = np.empty(S, dtype=np.int32)
        bs = self.kv.block_size
        max_blocks = 1
        cursor = 0
        for s, (req, c) in enumerate(zip(sb.reqs, sb.num_new_tokens)):
            start = req.num_computed_tokens
            table = self.kv.block_tables[req.req_id]
            max_blocks = max(max_blocks, len(table))
            for j in range(c):
                pos = start + j
                # async decode: the row's last token is still in flight
                # (pending tensor overrides fb.input_ids); placeholder here
                input_ids[cursor] = (req.token_at(pos)
                                     if pos < req.total_len else 0)
                positions[cursor] = pos
                slots[cursor] = table[pos // bs] * bs + pos % bs
                cursor += 1
            qlocs[s + 1] = cursor
            seq_lens[s] = start + c
        block_tables = np.zeros((S, max_blocks), dtype=np.int32)
        for s, req in enumerate(sb.reqs):
            t = self.kv.block_tables[req.req_id]
            block_tables[s, : len(t)] = t

        # rows that need logits: completing prefills + all decodes
        logits_rows: List[int] = []
        for s in range(sb.num_prefills):
            req, c = sb.reqs[s], sb.num_new_tokens[s]
            if req.num_computed_tokens + c == req.num_prompt_tokens:
                logits_rows.append(int(qlocs[s + 1]) - 1)
        for s in range(sb.num_prefills, S):
            logits_rows.append(int(qlocs[s + 1]) - 1)

        # prefill q-tile map (32 rows per tile) for the GPU flash kernel
        tiles_s: List[int] = []
        tiles_q0: List[int] = []
        for s in range(sb.num_prefills):
            c = sb.num_new_tokens[s]
            for j in range(0, c, 32):
                tiles_s.append(s)
                tiles_q0.append(j)
        prefill_token_count = int(qlocs[sb.num_prefills])

        dev = self.device
        return ForwardBatch(
            input_ids=torch.from_numpy(input_ids).to(dev),
            positions=torch.from_numpy(positions).to(dev),
            slot_mapping=torch.from_numpy(slots).to(dev),
            block_tables=torch.from_numpy(block_tables).to(dev),
            seq_lens=torch.from_numpy(seq_lens).to(dev),
            query_start_locs=torch.from_numpy(qlocs).to(dev),
            num_decodes_tail=sb.num_decodes,
            logits_idx=torch.tensor(logits_rows, dtype=torch.long, device=dev),
            max_seq_len=int(seq_lens.max()) if S else 0,
            max_query_len=int(max(sb.num_new_tokens)) if S else 0,
            tile_seq=torch.tensor(tiles_s, dtype=torch.int32, device=dev),
            tile_q0=torch.tensor(tiles_q0, dtype=torch.int32, device=dev),
            prefill_token_count=prefill_token_count,
        )

    def _sampling_reqs(self, sb: ScheduledBatch) -> List[Request]:
        out = []
        for s in range(sb.num_prefills):
            req, c = sb.reqs[s], sb.num_new_tokens[s]
            if req.num_computed_tokens + c == req.num_prompt_tokens:
                out.append(req)
        out.extend(sb.reqs[sb.num_prefills:])
        return out

    def _consume_pending(self, stats: StepStats) -> None:
        """Apply the previous async step's tokens (host side of the one-step
        lag). Skips rows that finished or were preempt-restarted meanwhile."""
        p = self._pending_decode
        if p is None:
            return
        self._pending_decode = None
        reqs, toks_t, lps_t, gens = p
        toks = toks_t.tolist()  # the only device wait on the async path —
        lps = lps_t.tolist()    # by now the GPU finished this step long ago
        for req, tok, lp, gen in zip(reqs, toks, lps, gens):
            if req.finish_reason is not None or req.alloc_gen != gen:
                continue
            self._apply_sampled(req, int(tok), float(lp), stats)

    def _async_eligible(self, sub: ScheduledBatch) -> bool:
        if not (self.cfg.async_decode and not self.spec.embedding
                and sub.num_prefills == 0 and sub.reqs):
            return False
        if any(r.fsm_id is not None for r in sub.reqs):
            return False  # the FSM mask needs the sampled token NOW
        p = self._pending_decode
        if p is None:
            return True
        # the lagged token tensor feeds this step's ids verbatim, so the
        # decode composition (and order) must be unchanged
        return len(p[0]) == len(sub.reqs) and all(
            a is b for a, b in zip(p[0], sub.reqs))

    @staticmethod
    def _drop_finished(sub: ScheduledBatch) -> ScheduledBatch:
        kept = [(r, c) for r, c in zip(sub.reqs, sub.num_new_tokens)
                if r.finish_reason is None]
        return ScheduledBatch(
            reqs=[r for r, _ in kept],
            num_new_tokens=[c for _, c in kept],
            num_prefills=sum(1 for r, _ in kept if r.in_prefill))

    @torch.no_grad()
    def step(self) -> StepStats:
        sb = self.scheduler.schedule()
        stats = StepStats()
        if not sb.reqs:
            # drain the final lagged tokens (finished-row extras only)
            self._consume_pending(stats)
            self.total_output_tokens += stats.output_tokens
            return stats
        stats.scheduled_tokens = sb.total_tokens

        # Split mixed steps: decodes replay a hipGraph, prefills run eager.
        sub_batches: List[ScheduledBatch] = []
        if (self.graph_runner is not None and sb.num_decodes > 0):
            if sb.num_prefills > 0:
                sub_batches.append(ScheduledBatch(
                    reqs=sb.reqs[: sb.num_prefills],
                    num_new_tokens=sb.num_new_tokens[: sb.num_prefills],
                    num_prefills=sb.num_prefills))
            sub_batches.append(ScheduledBatch(
                reqs=sb.reqs[sb.num_prefills:],
                num_new_tokens=[1] * sb.num_decodes,
                num_prefills=0))
        else:
            sub_batches.append(sb)

        async_mode = (len(sub_batches) == 1
                      and self._async_eligible(sub_batches[0]))
        if not async_mode and self._pending_decode is not None:
            # sync point: consume first; it may finish rows already scheduled
            # into this batch — drop them before executing
            self._consume_pending(stats)
            sub_batches = [self._drop_finished(s) for s in sub_batches]
            sub_batches = [s for s in sub_batches if s.reqs]

        if async_mode:
            sub = sub_batches[0]
            pend = self._pending_decode
            ids_t = pend[1] if pend is not None else None
            if self.graph_runner is not None and self.graph_runner.can_run(sub):
                logits = self.graph_runner.run(sub, ids_override=ids_t)
            else:
                fb = self._build_forward_batch(sub)
                if ids_t is not None:
                    fb.input_ids = ids_t
                hidden = self.model(fb, self.kv)
                logits = self.model.compute_logits(hidden[fb.logits_idx])
            steps = np.fromiter((r.num_computed_tokens + 1 for r in sub.reqs),
                                np.uint64, len(sub.reqs))
            toks_t, lps_t = self.sampler.sample(
                logits, list(sub.reqs), None, steps_override=steps,
                return_tensors=True)
            self._advance_computed(sub)
            new_pending = (list(sub.reqs), toks_t, lps_t,
                           [r.alloc_gen for r in sub.reqs])
            self._consume_pending(stats)  # previous step's tokens (lagged)
            self._pending_decode = new_pending
            self.total_prompt_tokens += stats.prefill_tokens
            self.total_output_tokens += stats.output_tokens
            return stats

        for sub in sub_batches:
            if (self.graph_runner is not None and self.graph_runner.can_run(sub)):
                logits = self.graph_runner.run(sub)
                sample_reqs = list(sub.reqs)
            elif (self.prefill_graph is not None
                  and self.prefill_graph.can_run(sub)):
                hidden = self.prefill_graph.run(sub)
                if self.spec.embedding:
                    self._embedding_update(sub, hidden, stats)
                    self._advance_computed(sub)
                    stats.prefill_tokens += sub.total_tokens
                    continue
                sample_reqs = self._sampling_reqs(sub)
                if sample_reqs:
                    rows, cum = [], 0
                    for req, c in zip(sub.reqs, sub.num_new_tokens):
                        cum += c
                        if req.num_computed_tokens + c == req.num_prompt_tokens:
                            rows.append(cum - 1)
                    idx = torch.tensor(rows, dtype=torch.long,
                                       device=hidden.device)
                    logits = self.model.compute_logits(hidden[idx])
                else:
                    logits = None
            else:
                fb = self._build_forward_batch(sub)
                hidden = self.model(fb, self.kv)
                if self.spec.embedding:
                    self._embedding_update(sub, hidden, stats)
                    self._advance_computed(sub)
                    stats.prefill_tokens += sub.total_tokens
                    continue
                sample_reqs = self._sampling_reqs(sub)
                logits = (self.model.compute_logits(hidden[fb.logits_idx])
                          if sample_reqs else None)
            if sample_reqs:
                fsm_mask = self._fsm_masks(sample_reqs, logits.device)
                tokens, lps = self.sampler.sample(logits, sample_reqs, fsm_mask)
            else:
                tokens, lps = [], []
            self._advance_computed(sub)
            for req, tok, lp in zip(sample_reqs, tokens, lps):
                self._apply_sampled(req, int(tok), float(lp), stats)
            stats.prefill_tokens += sum(sub.num_new_tokens[: sub.num_prefills])

        self.total_prompt_tokens += stats.prefill_tokens
        self.total_output_tokens += stats.output_tokens
        return stats

    def _advance_computed(self, sb: ScheduledBatch) -> None:
        for req, c in zip(sb.reqs, sb.num_new_tokens):
            req.num_computed_tokens += c

    def _fsm_masks(self, reqs: List[Request], device) -> Optional[torch.Tensor]:
        """Packed FSM mask rows [n, W] int32 (guided.py layout), or None when
        no row is guided. Mixed batches (concurrent jobs with and without
        schemas) group rows per FSM: one gather per FSM + one full-alive row
        for the unguided rows."""
        fsm_ids = {r.fsm_id for r in reqs}
        if fsm_ids == {None}:
            return None
        if None not in fsm_ids and len(fsm_ids) == 1:
            # homogeneous guided batch (the structured-job case): one gather
            fsm = self._fsms[next(iter(fsm_ids))]
            return fsm.mask_rows([r.fsm_state for r in reqs])
        from .guided import full_mask_row

        vocab = self.tokenizer.vocab_size
        if self._full_mask_row is None or self._full_mask_row.device != torch.device(device):
            self._full_mask_row = full_mask_row(vocab, device)
        W = self._full_mask_row.shape[0]
        mask = torch.empty((len(reqs), W), dtype=torch.int32, device=device)
        by_fsm: Dict[Optional[int], List[int]] = {}
        for i, r in enumerate(reqs):
            by_fsm.setdefault(r.fsm_id, []).append(i)
        for fid, rows in by_fsm.items():
            idx = torch.tensor(rows, dtype=torch.long, device=device)
            if fid is None:
                mask[idx] = self._full_mask_row
            else:
                mask[idx] = self._fsms[fid].mask_rows(
                    [reqs[i].fsm_state for i in rows]).to(device)
        return mask

    def _apply_sampled(self, req: Request, tok: int, lp: float, stats: StepStats) -> None:
        sp = req.sampling
        # FSM-guided rows honor ONLY EOS (the mask restricts it to schema-
        # accepting states): a user stop token/string landing mid-structure
        # would truncate the JSON and break the schema-validity guarantee
        # the templates rely on (found by deep property fuzzing)
        stop_hit = (tok == EOS_ID if req.fsm_id is not None
                    else tok in req.stop_ids(EOS_ID))
        if stop_hit:
            self.scheduler.finish(req, FinishReason.STOP)
            stats.finished.append(req)
            return
        req.output_token_ids.append(tok)
        req.cumulative_logprob += lp
        stats.output_tokens += 1
        if sp.stop and req.fsm_id is None:
            # BPE tokens are multi-byte: a stop string can complete (or even
            # be strictly inside) the newest token, so scan a decoded tail
            # window and trim at TEXT level (req.text_override)
            win = max(64, max(len(s) for s in sp.stop) + 8)
            out = self.tokenizer.decode(req.output_token_ids[-win:])
            for s in sp.stop:
                if s and s in out:
                    full = self.tokenizer.decode(req.output_token_ids)
                    # first occurrence == the one just completed (earlier
                    # ones would have finished the row on a previous step)
                    pos = full.find(s)
                    req.text_override = full[:pos] if pos >= 0 else full
                    self.scheduler.finish(req, FinishReason.STOP)
                    stats.finished.append(req)
                    return
        if req.fsm_id is not None:
            fsm = self._fsms[req.fsm_id]
            req.fsm_state = fsm.advance(req.fsm_state, tok)
            if req.fsm_state < 0:
                # defensive: a token escaped the mask (cannot happen while
                # FSM support is non-empty; guards a corrupted state from
                # ever reaching mask_rows where it would KeyError)
                self.scheduler.finish(req, FinishReason.STOP)
                stats.finished.append(req)
                return
            if fsm.must_stop(req.fsm_state):
                self.scheduler.finish(req, FinishReason.STOP)
                stats.finished.append(req)
                return
        if len(req.output_token_ids) >= sp.max_tokens or req.total_len >= self.cfg.max_model_len:
            self.scheduler.finish(req, FinishReason.LENGTH)
            stats.finished.append(req)

    # ---- embedding mode ----

    def _embedding_update(self, sb: ScheduledBatch, hidden: torch.Tensor,
                          stats: StepStats) -> None:
        # batched per-sequence chunk sums (no per-seq host sync)
        S = len(sb.reqs)
        counts = torch.tensor(sb.num_new_tokens, device=hidden.device)
        seg = torch.repeat_interleave(
            torch.arange(S, device=hidden.device), counts)
        sums = torch.zeros(S, hidden.shape[-1], dtype=torch.float32,
                           device=hidden.device)
        sums.index_add_(0, seg, hidden.float())
        done_rows: List[int] = []
        done_reqs: List[Request] = []
        for s, (req, c) in enumerate(zip(sb.reqs, sb.num_new_tokens)):
            completing = req.num_computed_tokens + c == req.num_prompt_tokens
            if req.req_id in self._emb_sums or not completing:
                if req.req_id in self._emb_sums:
                    self._emb_sums[req.req_id] += sums[s]
                else:
                    self._emb_sums[req.req_id] = sums[s].clone()
            if completing:
                done_rows.append(s)
                done_reqs.append(req)
        if done_reqs:
            vecs = []
            for s, req in zip(done_rows, done_reqs):
                v = self._emb_sums.pop(req.req_id, None)
                if v is None:
                    v = sums[s]
                vecs.append(v / max(1, req.num_prompt_tokens))
            mat = torch.stack(vecs)
            mat = mat / (mat.norm(dim=-1, keepdim=True) + 1e-12)
            mat_cpu = mat.cpu().numpy()  # ONE sync for the whole step
            for i, req in enumerate(done_reqs):
                self.embeddings[req.req_id] = mat_cpu[i]
                self.scheduler.finish(req, FinishReason.STOP)
                stats.finished.append(req)

    def output_text(self, req: Request) -> str:
        """Final output text of a finished request (stop-string trims are
        text-level because BPE trim points need not align to tokens)."""
        if req.text_override is not None:
            return req.text_override
        return self.tokenizer.decode(req.output_token_ids)

    # ---- convenience: run a list of prompts to completion (tests/smoke) ----

    def generate(
        self,
        prompts: List[str],
        sampling: Optional[SamplingParams] = None,
        system_prompt: Optional[str] = None,
        schema: Optional[dict] = None,
    ) -> List[str]:
        fsm_id = self.register_fsm(schema) if schema else None
        reqs = []
        for i, p in enumerate(prompts):
            ids = self.tokenizer.render_prompt(p, system_prompt)
            sp = sampling or SamplingParams(max_tokens=32, temperature=0.8)
            reqs.append(self.add_request(ids, sp, fsm_id=fsm_id, arrival_idx=i))
        while self.has_work():
            self.step()
        return [self.output_text(r) for r in reqs]
