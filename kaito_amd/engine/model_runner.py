"""Model execution: input prep, KV cache allocation, hipGraph decode capture.

MI355X-first specifics:
  * KV pool sized from hipMemGetInfo free-VRAM probe × gpu_memory_utilization
    (the reference's "--max-model-len=auto" / gpu-memory-utilization knobs,
    pkg/model/interface.go:308-312 + inference_api.py:439-496).
  * decode steps are captured into hipGraphs per batch-size bucket
    (the launch-bound inner loop: ~300 kernel launches per 8B-model step).
  * 288 GB HBM3E: default 0.90 utilization leaves >100 GB of KV for 8B.
"""
from __future__ import annotations

import logging
from pathlib import Path
from typing import Dict, List, Optional, Tuple

import torch

# hipBLASLt algo selections tuned offline on MI355X (torch TunableOp);
# shipped in-tree so every engine start gets the tuned GEMM kernels without
# paying the tuning cost. Regenerate with: python bench.py --tune-gemms.
TUNABLE_FILE = Path(__file__).resolve().parent.parent / "ops" / "tunableop_gfx950.csv"

from .config import EngineConfig
from .scheduler import ScheduledBatch
from .sequence import Sequence
from . import lora as lora_mod
from ..models.llama import AttnMetadata, LlamaForCausalLM
from ..parallel import state as ps

logger = logging.getLogger(__name__)


class ModelRunner:
    def __init__(self, cfg: EngineConfig):
        self.cfg = cfg
        self.device = torch.device(cfg.device)
        self.is_gpu = self.device.type == "cuda"
        torch.manual_seed(cfg.seed)
        self.model = LlamaForCausalLM(cfg.model).to(self.device)
        self.kv_caches: List[Tuple[torch.Tensor, torch.Tensor]] = []
        self.num_gpu_blocks = 0
        self.max_model_len = cfg.max_model_len or cfg.model.max_position
        self._graphs: Dict[int, tuple] = {}
        self._graph_pool = None
        self.lora_manager = None
        if cfg.enable_lora:
            self.lora_manager = lora_mod.LoRAManager(
                self.model, cfg.max_loras, cfg.max_lora_rank,
                device=cfg.device, dtype=cfg.model.dtype)

    # ------------------------------------------------------------- setup
    def load_model(self, weights_path: Optional[str] = None, seed: int = 0):
        awq_ckpt = False
        if weights_path:
            import json
            import os
            cfg_json = os.path.join(weights_path, "config.json")
            if os.path.exists(cfg_json):
                with open(cfg_json) as f:
                    qm = (json.load(f).get("quantization_config") or {}) \
                        .get("quant_method", "")
                awq_ckpt = qm == "awq"
            from ..models.loader import load_safetensors_weights
            load_safetensors_weights(self.model, weights_path,
                                     skip_projections=awq_ckpt)
        else:
            self.model.random_init(seed)
        if awq_ckpt:
            from ..models.quant import load_awq_checkpoint
            n = load_awq_checkpoint(self.model, weights_path)
            logger.info("loaded %d AWQ-quantized linears", n)
        elif self.cfg.model.quant_method in ("awq", "w4a16"):
            from ..models.quant import quantize_parallel_linears
            n = quantize_parallel_linears(self.model)
            logger.info("quantized %d linears to W4A16", n)
        self.model.init_rope(self.device, self.max_model_len)
        self.model.eval()
        return self

    def profile_and_allocate_kv(self) -> int:
        """Size the KV pool from free VRAM (auto max-model-len semantics)."""
        cfg = self.cfg
        m = cfg.model
        tp = max(cfg.tensor_parallel_size, 1)
        kvh = max(m.num_kv_heads // tp, 1)
        fp8 = cfg.kv_cache_dtype == "fp8"
        if fp8 and not self.is_gpu:
            raise ValueError("kv_cache_dtype=fp8 needs a GPU")
        if fp8 and m.is_mla:
            raise ValueError("fp8 KV cache not supported for MLA models")
        elem = 1 if fp8 else 2
        if m.is_mla:
            # ONE compressed latent row per token (c_kv ‖ k_rope), shared
            # by all heads and replicated across TP ranks (models/mla.py)
            block_bytes = m.num_layers * cfg.block_size * m.kv_cache_row \
                * elem
        else:
            block_bytes = 2 * m.num_layers * kvh * cfg.block_size \
                * m.head_dim * elem
        if cfg.num_gpu_blocks is not None:
            n_blocks = cfg.num_gpu_blocks
        elif self.is_gpu:
            free, total = torch.cuda.mem_get_info(self.device)
            budget = int(total * cfg.gpu_memory_utilization) - (total - free)
            n_blocks = max(budget // block_bytes, 16)
        else:
            n_blocks = 512  # CPU tests
        # cap: no point holding more than max_num_seqs * max_model_len
        cap = cfg.max_num_seqs * ((self.max_model_len + cfg.block_size - 1)
                                  // cfg.block_size) + 1
        n_blocks = min(n_blocks, cap)
        self.num_gpu_blocks = n_blocks
        kvs = []
        kv_dtype = torch.uint8 if fp8 else m.dtype
        for _ in range(self.model.num_local_layers):
            if m.is_mla:
                # aliased pair: the latent cache IS both k and v. One
                # spare block at the end absorbs writes for PADDING slots
                # (-1): the graph-safe index_copy_ cache write redirects
                # them there (models/mla.py) the way the reshape_and_cache
                # kernel skips negatives; the pool never hands it out and
                # no block table references it.
                c = torch.zeros(n_blocks + 1, cfg.block_size, m.kv_cache_row,
                                dtype=kv_dtype, device=self.device)
                kvs.append((c, c))
                continue
            k = torch.zeros(n_blocks, kvh, cfg.block_size, m.head_dim,
                            dtype=kv_dtype, device=self.device)
            v = torch.zeros_like(k)
            kvs.append((k, v))
        self.kv_caches = kvs
        logger.info("KV pool: %d blocks (%d tokens), %.2f GiB", n_blocks,
                    n_blocks * cfg.block_size,
                    n_blocks * block_bytes / (1 << 30))
        return n_blocks

    # ------------------------------------------------------------- helpers
    def _slot(self, seq: Sequence, pos: int) -> int:
        bs = self.cfg.block_size
        return seq.block_table[pos // bs] * bs + pos % bs

    # ------------------------------------------------------------- prefill
    @torch.no_grad()
    def execute_prefill(self, chunks) -> torch.Tensor:
        """Run a (possibly chunked) prefill batch. Each chunk covers prompt
        tokens [start, start+length) of its sequence; chunks with start>0
        or incomplete prompts use context attention over the paged cache.
        Returns hidden states of the LAST token of each COMPLETING chunk,
        in chunk order (aligned with batch.sampling_seqs)."""
        ids, pos, slots, cu = [], [], [], [0]
        fresh = all(c.start == 0 and c.completes for c in chunks)
        for c in chunks:
            toks = c.seq.context_token_ids[c.start:c.start + c.length]
            ids.extend(toks)
            pos.extend(range(c.start, c.start + c.length))
            slots.extend(self._slot(c.seq, p)
                         for p in range(c.start, c.start + c.length))
            cu.append(cu[-1] + c.length)
        dev = self.device
        input_ids = torch.tensor(ids, dtype=torch.long, device=dev)
        positions = torch.tensor(pos, dtype=torch.long, device=dev)
        kv_lens = None
        block_tables = None
        if not fresh:
            import numpy as np
            mb = max(len(c.seq.block_table) for c in chunks)
            flat = np.zeros((len(chunks), mb), dtype=np.int32)
            for i, c in enumerate(chunks):
                flat[i, :len(c.seq.block_table)] = c.seq.block_table
            block_tables = torch.from_numpy(flat).to(dev)
            kv_lens = torch.tensor([c.start + c.length for c in chunks],
                                   dtype=torch.int32, device=dev)
        meta = AttnMetadata(
            is_prefill=True,
            slot_mapping=torch.tensor(slots, dtype=torch.long, device=dev),
            cu_seqlens=torch.tensor(cu, dtype=torch.int32, device=dev),
            max_seqlen=max(c.length for c in chunks),
            block_tables=block_tables,
            kv_lens=kv_lens)
        if self.lora_manager is not None:
            tok_ids = []
            for c in chunks:
                tok_ids.extend([c.seq.lora_id] * c.length)
            lora_mod.set_active(self.lora_manager, torch.tensor(
                tok_ids, dtype=torch.int32, device=dev))
        hidden = self._model_forward(input_ids, positions, meta)
        if self.lora_manager is not None:
            lora_mod.set_active(None, None)
        if hidden is None:
            return None  # intermediate PP stage
        last = [cu[i + 1] - 1 for i, c in enumerate(chunks) if c.completes]
        if not last:
            return hidden[:0]
        return hidden[torch.tensor(last, device=dev)]

    # ------------------------------------------------------------- decode
    def _init_decode_buffers(self):
        cfg = self.cfg
        max_bs = max(max(cfg.graph_batch_sizes), cfg.max_num_seqs)
        mb = cfg.max_blocks_per_seq(self.max_model_len)
        dev = self.device
        self._buf = {
            "input_ids": torch.zeros(max_bs, dtype=torch.long, device=dev),
            "positions": torch.zeros(max_bs, dtype=torch.long, device=dev),
            "slot_mapping": torch.full((max_bs,), -1, dtype=torch.long, device=dev),
            "block_tables": torch.zeros(max_bs, mb, dtype=torch.int32, device=dev),
            "seq_lens": torch.zeros(max_bs, dtype=torch.int32, device=dev),
            "lora_ids": torch.full((max_bs,), -1, dtype=torch.int32, device=dev),
        }
        self._max_blocks = mb

    def _model_forward(self, input_ids, positions, meta):
        """PP-aware forward: non-first stages receive the previous stage's
        hidden; non-last stages send theirs on. Returns the final hidden on
        the LAST stage, None elsewhere."""
        st = ps.get_state()
        if st.pp_size == 1:
            return self.model(input_ids, positions, self.kv_caches, meta)
        T = input_ids.shape[0]
        H = self.cfg.model.hidden_size
        if st.is_first_stage:
            out = self.model(input_ids, positions, self.kv_caches, meta)
        else:
            hin = ps.pp_recv_prev((T, H), self.cfg.model.dtype, self.device)
            out = self.model(input_ids, positions, self.kv_caches, meta,
                             hidden_in=hin)
        if not st.is_last_stage:
            ps.pp_send_next(out)
            return None
        return out

    def _decode_forward(self, bs: int) -> torch.Tensor:
        b = self._buf
        if self.lora_manager is not None:
            lora_mod.set_active(self.lora_manager, b["lora_ids"][:bs])
        meta = AttnMetadata(
            is_prefill=False,
            slot_mapping=b["slot_mapping"][:bs],
            block_tables=b["block_tables"][:bs],
            seq_lens=b["seq_lens"][:bs])
        return self._model_forward(b["input_ids"][:bs], b["positions"][:bs],
                                   meta)

    def setup_tunable(self):
        """Enable TunableOp lookups from the in-tree MI355X tuning cache
        (hipBLASLt's default heuristic picks catastrophically bad kernels
        for some skinny decode shapes — measured 557us vs 44us for
        near-identical work, profiles/r01_decode_profile.md)."""
        if not self.is_gpu:
            return
        import torch.cuda.tunable as tunable
        tunable.enable(True)
        tunable.tuning_enable(False)
        if TUNABLE_FILE.exists():
            tunable.read_file(str(TUNABLE_FILE))

    def tune_gemms(self, out_file: Optional[str] = None):
        """Tune hipBLASLt algo selection for every decode-bucket GEMM shape
        (+ one full prefill shape) and persist the results."""
        if not self.is_gpu:
            return
        import torch.cuda.tunable as tunable
        dest = out_file or str(TUNABLE_FILE)
        tunable.enable(True)
        if TUNABLE_FILE.exists():
            tunable.read_file(str(TUNABLE_FILE))
        # results are flushed to the filename at process exit
        tunable.set_filename(dest)
        tunable.tuning_enable(True)
        if not hasattr(self, "_buf"):
            self._init_decode_buffers()
        self._buf["seq_lens"].fill_(1)
        for bs in sorted(self.cfg.graph_batch_sizes, reverse=True):
            if bs > self.cfg.max_num_seqs:
                continue
            hidden = self._decode_forward(bs)
            self.model.compute_logits(hidden)
            torch.cuda.synchronize()
        self._buf["seq_lens"].zero_()
        # prefill shapes: pure-prefill steps pin M at max_num_batched_tokens;
        # mixed (overlapped) steps run chunks around mixed_prefill_tokens.
        # (the untuned default heuristic costs ~1/3 of bench time there)
        dev = self.device
        mcfg = self.cfg.model
        from ..models.llama import AttnMetadata
        mp = self.cfg.mixed_prefill_tokens
        shapes = {self.cfg.max_num_batched_tokens, mp, 4096, 2000, 1024,
                  512, 256}
        for M in sorted(shapes, reverse=True):
            step = min(M, 256)
            cu = list(range(0, M, step)) + [M]   # covers a non-multiple tail
            meta = AttnMetadata(
                is_prefill=True,
                slot_mapping=torch.full((M,), -1, dtype=torch.long,
                                        device=dev),
                cu_seqlens=torch.tensor(cu, dtype=torch.int32, device=dev),
                max_seqlen=step)
            ids = torch.randint(0, mcfg.vocab_size, (M,), device=dev)
            pos = torch.arange(M, device=dev) % step
            hidden = self.model(ids, pos, None, meta)
            self.model.compute_logits(hidden[:64])
            torch.cuda.synchronize()
        tunable.tuning_enable(False)
        logger.info("TunableOp tuned; results flush to %s at exit", dest)

    def capture_decode_graphs(self):
        """Capture hipGraphs for each decode bucket (largest first so the
        shared memory pool is sized once)."""
        if not self.is_gpu or self.cfg.enforce_eager:
            return
        m = self.cfg.model
        tp = max(ps.get_state().tp_size, 1)
        ie = m.moe_intermediate_size or m.intermediate_size
        ie_local = ie if (tp > 1 and m.num_experts % tp == 0) or tp == 1 \
            else ie // tp   # mirrors MoEMLP's EP-vs-IE sharding choice
        if m.num_experts > 0 and (m.hidden_size % 64 != 0 or
                                  ie_local % 64 != 0):
            # odd shapes fall back to the per-expert torch loop, whose
            # host-side segment reads cannot be captured
            logger.info("MoE model with non-64-aligned dims: "
                        "skipping decode graph capture")
            return
        if not hasattr(self, "_buf"):
            self._init_decode_buffers()
        self._buf["seq_lens"].fill_(1)  # benign shapes for capture
        torch.cuda.synchronize()
        for bs in sorted(self.cfg.graph_batch_sizes, reverse=True):
            if bs > self.cfg.max_num_seqs:
                continue
            # warmup (allocator settles) then capture
            self._decode_forward(bs)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, pool=self._graph_pool):
                out = self._decode_forward(bs)
            if self._graph_pool is None:
                self._graph_pool = g.pool()
            self._graphs[bs] = (g, out)
        self._buf["seq_lens"].zero_()
        torch.cuda.synchronize()
        logger.info("captured %d decode hipGraphs", len(self._graphs))

    def _graph_bucket(self, bs: int) -> Optional[int]:
        for b in sorted(self._graphs):
            if b >= bs:
                return b
        return None

    @torch.no_grad()
    def execute_decode(self, seqs: List[Sequence],
                       sampled: Optional[torch.Tensor] = None,
                       pending_map: Optional[Tuple[torch.Tensor, dict]] = None
                       ) -> torch.Tensor:
        """One token per sequence; returns hidden [B, H].

        Fast path: when the running set is unchanged since the previous
        decode step and `sampled` (last step's token ids, on device) is
        given, all input updates happen device-side — no host round-trip
        except dirty block-table rows (block-boundary crossings).

        Rebuild path: lengths come from seq.sched_tokens (pipelined steps may
        be unresolved); input token VALUES for sequences with an in-flight
        step are sourced device-side from `pending_map` = (tokens_tensor,
        {seq_id: index}).
        """
        if not hasattr(self, "_buf"):
            self._init_decode_buffers()
            self._last_ids: List[int] = []
        bs = len(seqs)
        b = self._buf
        ids_now = [s.seq_id for s in seqs]
        reuse = (sampled is not None and ids_now == self._last_ids)
        if reuse:
            b["input_ids"][:bs].copy_(sampled[:bs])
            b["positions"][:bs] += 1
            b["seq_lens"][:bs] += 1
            for i, s in enumerate(seqs):
                if getattr(s, "_bt_dirty", False):
                    n = len(s.block_table)
                    b["block_tables"][i, :n].copy_(torch.tensor(
                        s.block_table, dtype=torch.int32), non_blocking=True)
                    s._bt_dirty = False
            # slot = bt[i, pos // BS] * BS + pos % BS  (device-side gather)
            pos = b["positions"][:bs]
            blk = torch.gather(b["block_tables"][:bs].long(), 1,
                               (pos // self.cfg.block_size).unsqueeze(1)).squeeze(1)
            b["slot_mapping"][:bs].copy_(
                blk * self.cfg.block_size + pos % self.cfg.block_size)
        else:
            ptoks, pmap = pending_map if pending_map is not None else (None, {})
            ids, dst, src = [], [], []
            for i, s in enumerate(seqs):
                j = pmap.get(s.seq_id)
                if j is not None and s.sched_tokens > s.num_tokens:
                    ids.append(0)
                    dst.append(i)
                    src.append(j)
                else:
                    ids.append(s.last_token_id)
            pos = [s.sched_tokens - 1 for s in seqs]
            slots = [self._slot(s, p) for s, p in zip(seqs, pos)]
            lens = [s.sched_tokens for s in seqs]
            b["input_ids"][:bs].copy_(
                torch.tensor(ids, dtype=torch.long), non_blocking=True)
            if dst:
                didx = torch.tensor(dst, dtype=torch.long, device=self.device)
                sidx = torch.tensor(src, dtype=torch.long, device=self.device)
                b["input_ids"][:bs].index_copy_(0, didx, ptoks[sidx])
            b["positions"][:bs].copy_(
                torch.tensor(pos, dtype=torch.long), non_blocking=True)
            b["slot_mapping"][:bs].copy_(
                torch.tensor(slots, dtype=torch.long), non_blocking=True)
            b["seq_lens"][:bs].copy_(
                torch.tensor(lens, dtype=torch.int32), non_blocking=True)
            import numpy as np
            flat = np.zeros((bs, self._max_blocks), dtype=np.int32)
            for i, s in enumerate(seqs):
                flat[i, :len(s.block_table)] = s.block_table
                s._bt_dirty = False
            b["block_tables"][:bs].copy_(
                torch.from_numpy(flat), non_blocking=True)
            if self.lora_manager is not None:
                b["lora_ids"][:bs].copy_(torch.tensor(
                    [s.lora_id for s in seqs], dtype=torch.int32),
                    non_blocking=True)
        self._last_ids = ids_now

        bucket = self._graph_bucket(bs) if self._graphs else None
        if bucket is not None:
            # zero the padded tail so padded rows do no work
            if bucket > bs:
                b["seq_lens"][bs:bucket].zero_()
                b["slot_mapping"][bs:bucket].fill_(-1)
            g, out = self._graphs[bucket]
            g.replay()
            return out[:bs]
        return self._decode_forward(bs)

    # ------------------------------------------------------------- step
    @torch.no_grad()
    def execute(self, batch: ScheduledBatch,
                sampled: Optional[torch.Tensor] = None,
                pending_map=None) -> torch.Tensor:
        """Run the batch; returns logits [B, vocab] for the last tokens."""
        if batch.is_prefill:
            hidden = self.execute_prefill(batch.chunks)
        else:
            hidden = self.execute_decode(batch.seqs, sampled, pending_map)
        if hidden is None:
            return None  # intermediate PP stage
        return self.model.compute_logits(hidden)
