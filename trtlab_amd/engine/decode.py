"""Incremental (KV-cache) decoding for GPT-2- and LLaMA-family models.

Beyond-reference capability (the CUDA reference served static TensorRT
engines only): a `DecodeSession` holds per-layer K/V caches resident in
HBM and replays ONE hipGraph per generated token. All position dependence
flows through a device-side counter (csrc/kernels/decode.hip), so the
captured graph needs no re-instantiation between steps:

    embed(ids, pos) -> [per layer: ln1 -> qkv gemm -> kv_append ->
    decode_attention -> proj gemm -> +res -> ln2 -> ff1+gelu -> ff2 ->
    +res] -> ln_f -> advance_pos

Caveat: the raw-op GEMMs share the lazily-grown module-level scratch
buffer (csrc/ext.cpp test_scratch); the captured graph bakes its address,
so other raw-op users must not force a regrow while a session is live
(engine-managed contexts are unaffected).

Weights come from a `build_gpt2(embeddings=True)` or `build_llama()` IR
graph (the same random-init builders the full-sequence engine uses, so
prefill/decode can be cross-checked). The recipe is detected from the
node names: `l0_rms1` present -> LLaMA (RMSNorm + RoPE + SwiGLU, no
biases; RoPE reads the device position counter, so the captured decode
graph stays position-free); otherwise GPT-2 (LayerNorm + learned pos
emb + GELU). prefill()/step()/verify_chunk() and the paged KV pool work
for both.
"""
from __future__ import annotations

from typing import Dict, List

import numpy as np



class PagedKVPool:
    """vLLM-style paged KV memory: ONE fixed physical pool per layer,
    shared by every slot, mapped through a per-slot device block table.
    Pages hold 64 positions x heads x 64 dims; a page id is valid across
    all layers (each layer has its own K/V pools, ids allocated in
    lockstep), so one table serves the whole stack. Host-side free list;
    idle/reset slots return their pages — parked sequences hold ZERO
    cache memory (the paged payoff vs the dense [B][H][smax][64] layout).
    """

    def __init__(self, layers: int, heads: int, batch: int,
                 num_pages: int, max_pages_per_slot: int,
                 device: int = 0, head_dim: int = 64):
        import torch

        from trtlab_amd import native

        self._C = native()
        self._torch = torch
        torch.cuda.set_device(device)
        self.layers = layers
        self.heads = heads
        self.batch = batch
        self.num_pages = num_pages
        self.max_pages = max_pages_per_slot
        self.head_dim = head_dim
        self.kpools = [torch.zeros(num_pages, 64, heads, head_dim,
                                   dtype=torch.half, device="cuda")
                       for _ in range(layers)]
        self.vpools = [torch.zeros_like(k) for k in self.kpools]
        self.table = torch.full((batch, max_pages_per_slot), -1,
                                dtype=torch.int32, device="cuda")
        self._table_host = np.full((batch, max_pages_per_slot), -1,
                                   np.int32)
        self._free = list(range(num_pages - 1, -1, -1))

    @property
    def pages_free(self) -> int:
        return len(self._free)

    def ensure(self, b: int, logical_idx: int) -> None:
        """Map slot b's logical page if unmapped (called by the session
        right before a step that will write position logical_idx*64+...)."""
        if logical_idx >= self.max_pages:
            raise RuntimeError(
                f"slot {b}: logical page {logical_idx} >= max_pages")
        if self._table_host[b, logical_idx] >= 0:
            return
        if not self._free:
            raise MemoryError(
                "PagedKVPool exhausted (all pages mapped) — free a slot")
        pid = self._free.pop()
        self._table_host[b, logical_idx] = pid
        ent = np.array([pid], np.int32)
        self._C.memory.memcpy_h2d(
            self.table.data_ptr() + (b * self.max_pages + logical_idx) * 4,
            ent, 4)

    def free_slot(self, b: int) -> int:
        """Return slot b's pages to the free list; returns count freed."""
        n = 0
        for i in range(self.max_pages):
            pid = int(self._table_host[b, i])
            if pid >= 0:
                self._free.append(pid)
                self._table_host[b, i] = -1
                n += 1
        row = np.full(self.max_pages, -1, np.int32)
        self._C.memory.memcpy_h2d(
            self.table.data_ptr() + b * self.max_pages * 4, row,
            row.nbytes)
        return n


class DecodeSession:
    """One generation session: fixed batch, growing position."""

    def __init__(self, graph, batch: int, smax: int = 1024, device: int = 0,
                 capture: bool = True, lm_head: bool = False,
                 fused: bool = False, paged=None):
        """fused=True (needs batch <= 64): EXPERIMENTAL horizontal kernel
        fusion for the latency-bound step — LN / residual-add / embed
        prologues and KV-scatter / GeLU epilogues fold into the small-M
        GEMMs (csrc decode_gemm_fused), cutting ~8 kernels/layer to 5.
        MEASURED on MI355X (profiles/README r2): the fused GEMMs lose
        split-K + the deep staging pipeline and only fill cdiv(N,64)
        workgroups, so the step is currently ~2x SLOWER than the unfused
        autotuned path (0.86 -> 1.63 ms b8) despite 40% fewer kernels —
        default OFF; numerics are verified (test_fused_decode_matches_
        unfused). Making the fused GEMM pipeline-deep is the round-3
        follow-up. The residual stream ping-pongs between two buffers
        (the fused ADD_LN writes the NEW stream while other blocks still
        read the old)."""
        import torch

        from trtlab_amd import native
        from trtlab_amd.engine.planner import (EPI_BIAS, EPI_BIAS_GELU,
                                               EPI_NONE)

        self._C = native()
        self._torch = torch
        torch.cuda.set_device(device)
        self.batch = batch
        self.smax = smax
        self.capture = capture
        self._epi_bias = EPI_BIAS
        self._epi_gelu = EPI_BIAS_GELU
        self._epi_none = EPI_NONE

        # ---- pull weights out of the IR graph by node kind/name ----
        # Two layer recipes share the session plumbing (KV caches, paged
        # pool, graph capture, position counters, spec-decode chunks):
        #   gpt2  — LayerNorm + learned pos emb + GELU MLP (l{i}_ln1 ...)
        #   llama — RMSNorm + RoPE + SwiGLU, no biases   (l{i}_rms1 ...)
        nodes = {n.name: n for n in graph.nodes}
        self.arch = "llama" if "l0_rms1" in nodes else "gpt2"
        emb = next(n for n in graph.nodes if n.kind == "embedding")
        self.hidden = emb.attrs["tok"].shape[1]
        self.vocab = emb.attrs["tok"].shape[0]
        assert emb.attrs["pos"].shape[0] >= 1

        def dev16(a):
            return torch.from_numpy(np.ascontiguousarray(a, np.float32)) \
                .half().cuda()

        def dev32(a):
            return torch.from_numpy(np.ascontiguousarray(a, np.float32)).cuda()

        self.tok = dev16(emb.attrs["tok"])
        self.posemb = dev16(emb.attrs["pos"])
        if self.posemb.shape[0] < smax:
            self.smax = smax = int(self.posemb.shape[0])

        self.layers: List[Dict] = []
        li = 0
        while f"l{li}_qkv" in nodes:
            att = nodes[f"l{li}_att"]
            self.heads = att.attrs["heads"]
            self.hd = att.attrs.get("head_dim",
                                    self.hidden // self.heads)
            if self.hd not in (64, 128):
                raise ValueError("decode: head_dim must be 64 or 128")
            lay = {}
            if self.arch == "llama":
                self.rms_eps = float(
                    nodes[f"l{li}_rms1"].attrs.get("eps", 1e-5))
                lay["rms1_g"] = dev32(nodes[f"l{li}_rms1"].attrs["gamma"])
                lay["rms2_g"] = dev32(nodes[f"l{li}_rms2"].attrs["gamma"])
                self.theta = float(
                    nodes[f"l{li}_rope"].attrs.get("theta", 10000.0))
                for key in ("qkv", "proj", "gate", "up", "down"):
                    lay[key + "_w"] = dev16(nodes[f"l{li}_{key}"]
                                            .attrs["weight"])
            else:
                for key, nm in (("ln1", f"l{li}_ln1"), ("ln2", f"l{li}_ln2")):
                    lay[key + "_g"] = dev32(nodes[nm].attrs["gamma"])
                    lay[key + "_b"] = dev32(nodes[nm].attrs["beta"])
                for key in ("qkv", "proj", "ff1", "ff2"):
                    n = nodes[f"l{li}_{key}"]
                    lay[key + "_w"] = dev16(n.attrs["weight"])
                    lay[key + "_b"] = dev32(n.attrs["bias"])
            lay["kcache"] = torch.zeros(batch, self.heads, smax, self.hd,
                                        dtype=torch.half, device="cuda")
            lay["vcache"] = torch.zeros_like(lay["kcache"])
            self.layers.append(lay)
            li += 1
        self.n_layers = li
        if self.arch == "llama":
            if self.hidden > 2048:
                raise ValueError("llama decode: hidden > 2048 (rmsnorm "
                                 "kernel row limit)")
            self.inter = self.layers[0]["gate_w"].shape[0]
            self.lnf_g = dev32(nodes["rms_f"].attrs["gamma"])
            self.lnf_b = None
        else:
            self.inter = self.layers[0]["ff1_w"].shape[0]
            gf = nodes["ln_f"]
            self.lnf_g = dev32(gf.attrs["gamma"])
            self.lnf_b = dev32(gf.attrs["beta"])
        self.lm_head = lm_head  # logits = h @ tok^T (weight tying)

        B, Hd = batch, self.hidden
        self.ids = torch.zeros(B, dtype=torch.int32, device="cuda")
        self.pos = torch.zeros(B, dtype=torch.int32, device="cuda")
        self._slot_steps = np.zeros(B, np.int64)
        self._active = np.ones(B, bool)  # host mirror of pos[b] >= 0
        self.h = torch.zeros(B, Hd, dtype=torch.half, device="cuda")
        self.x = torch.zeros(B, Hd, dtype=torch.half, device="cuda")
        self.qkv = torch.zeros(B, 3 * Hd, dtype=torch.half, device="cuda")
        self.att = torch.zeros(B, Hd, dtype=torch.half, device="cuda")
        self.x2 = torch.zeros(B, Hd, dtype=torch.half, device="cuda")
        self.ff = torch.zeros(B, self.inter, dtype=torch.half, device="cuda")
        # SwiGLU needs gate AND up live at once (silu(gate) * up)
        self.ff2 = (torch.zeros(B, self.inter, dtype=torch.half,
                                device="cuda")
                    if self.arch == "llama" else None)
        self.out = torch.zeros(B, Hd, dtype=torch.half, device="cuda")
        self.logits = (torch.zeros(B, self.vocab, dtype=torch.half,
                                   device="cuda") if lm_head else None)
        # device-side greedy head: argmax ids land here every step (the
        # argmax kernel is captured in the step graph; step(return_ids=
        # True) then moves B ints instead of B*vocab logits over PCIe)
        self.gids = (torch.zeros(B, dtype=torch.int32, device="cuda")
                     if lm_head else None)
        self.supports_ids = lm_head
        self.h2 = torch.zeros(B, Hd, dtype=torch.half, device="cuda")
        # paged KV mode: paged = a PagedKVPool (shared with other
        # sessions) or True (private pool sized for smax). The dense
        # per-layer caches above are then released — slots draw pages on
        # demand and return them on idle/reset.
        self.kv_pool = None
        if paged:
            mp = (smax + 63) // 64
            self.kv_pool = paged if isinstance(paged, PagedKVPool) else \
                PagedKVPool(self.n_layers, self.heads, B,
                            num_pages=B * mp, max_pages_per_slot=mp,
                            device=device, head_dim=self.hd)
            if self.kv_pool.layers != self.n_layers or \
                    self.kv_pool.heads != self.heads or \
                    getattr(self.kv_pool, "head_dim", 64) != self.hd:
                raise ValueError("pool layer/head mismatch")
            for lay in self.layers:  # release the dense caches
                lay["kcache"] = None
                lay["vcache"] = None
        self.fused = bool(fused)
        if self.fused and self.arch != "gpt2":
            raise ValueError("fused decode supports the gpt2 recipe only")
        if self.fused and (B > 64 or not lm_head):
            raise ValueError("fused decode needs batch <= 64 and lm_head")
        if self.fused and self.hd != 64:
            raise ValueError("fused decode supports head_dim 64 only")
        if self.fused and self.kv_pool is not None:
            raise ValueError("fused + paged not supported together yet")
        self.stream = self._C.hip.stream_create()
        self._graph = 0
        self._steps = 0

    # ------------------------------------------------------------ plumbing
    def _enqueue_fused(self):
        """Fused decode step: per layer [qkv(ADD_LN/EMBED_LN prologue +
        KV-scatter epilogue), attention, proj, ff1(ADD_LN + GeLU), ff2]
        + fused lm head — ~5 kernels/layer instead of 8."""
        C, s = self._C, self.stream
        B, Hd = self.batch, self.hidden
        ops = C.ops
        scale = 1.0 / float(np.sqrt(float(self.hd)))
        h_cur, h_nxt = self.h, self.h2
        for li, lay in enumerate(self.layers):
            if li == 0:  # embed + ln1 + qkv + kv scatter, persists h
                ops.decode_gemm_fused(
                    3, 2, x=0, r=0, h_out=h_cur.data_ptr(),
                    gamma=lay["ln1_g"].data_ptr(),
                    beta=lay["ln1_b"].data_ptr(),
                    B=lay["qkv_w"].data_ptr(), bias=lay["qkv_b"].data_ptr(),
                    C=self.qkv.data_ptr(), ids=self.ids.data_ptr(),
                    tok=self.tok.data_ptr(), posemb=self.posemb.data_ptr(),
                    pos=self.pos.data_ptr(),
                    kcache=lay["kcache"].data_ptr(),
                    vcache=lay["vcache"].data_ptr(), M=B, N=3 * Hd, K=Hd,
                    heads=self.heads, smax=self.smax, stream=s, sync=False)
            else:  # h_nxt = h_cur + ff2_out; qkv = ln1(h_nxt) @ W (+kv)
                ops.decode_gemm_fused(
                    2, 2, x=self.x2.data_ptr(), r=h_cur.data_ptr(),
                    h_out=h_nxt.data_ptr(), gamma=lay["ln1_g"].data_ptr(),
                    beta=lay["ln1_b"].data_ptr(),
                    B=lay["qkv_w"].data_ptr(), bias=lay["qkv_b"].data_ptr(),
                    C=self.qkv.data_ptr(), pos=self.pos.data_ptr(),
                    kcache=lay["kcache"].data_ptr(),
                    vcache=lay["vcache"].data_ptr(), M=B, N=3 * Hd, K=Hd,
                    heads=self.heads, smax=self.smax, stream=s, sync=False)
                h_cur, h_nxt = h_nxt, h_cur
            ops.decode_attention(self.qkv.data_ptr(),
                                 lay["kcache"].data_ptr(),
                                 lay["vcache"].data_ptr(),
                                 self.att.data_ptr(), self.pos.data_ptr(),
                                 B, self.heads, self.smax, scale, stream=s,
                                 sync=False)
            ops.gemm_bt(0, self.att.data_ptr(), lay["proj_w"].data_ptr(),
                        self.x2.data_ptr(), bias=lay["proj_b"].data_ptr(),
                        M=B, N=Hd, K=Hd, epi=self._epi_bias, stream=s,
                        sync=False)
            # h_nxt = h_cur + proj_out; ff = gelu(ln2(h_nxt) @ W1)
            ops.decode_gemm_fused(
                2, 1, x=self.x2.data_ptr(), r=h_cur.data_ptr(),
                h_out=h_nxt.data_ptr(), gamma=lay["ln2_g"].data_ptr(),
                beta=lay["ln2_b"].data_ptr(), B=lay["ff1_w"].data_ptr(),
                bias=lay["ff1_b"].data_ptr(), C=self.ff.data_ptr(),
                pos=self.pos.data_ptr(), M=B, N=self.inter, K=Hd, stream=s,
                sync=False)
            h_cur, h_nxt = h_nxt, h_cur
            ops.gemm_bt(0, self.ff.data_ptr(), lay["ff2_w"].data_ptr(),
                        self.x2.data_ptr(), bias=lay["ff2_b"].data_ptr(),
                        M=B, N=Hd, K=self.inter, epi=self._epi_bias,
                        stream=s, sync=False)
        # head stays UNFUSED: at N=vocab the fused kernel's per-block LN
        # stats replicate across ~786 blocks (measured 102.6 vs 14.5 us,
        # tools/dbg_fused_perf) — add_layernorm + plain gemm_bt win there
        ops.add_layernorm(0, self.x2.data_ptr(), h_cur.data_ptr(),
                          self.lnf_g.data_ptr(), self.lnf_b.data_ptr(),
                          self.out.data_ptr(), M=B, N=Hd, stream=s,
                          sync=False)
        ops.gemm_bt(0, self.out.data_ptr(), self.tok.data_ptr(),
                    self.logits.data_ptr(), M=B, N=self.vocab, K=Hd,
                    epi=self._epi_none, stream=s, sync=False)
        ops.argmax_rows(self.logits.data_ptr(), self.gids.data_ptr(),
                        B, self.vocab, stream=s, sync=False)
        # no cross-step residual state: layer 0's EMBED prologue
        # regenerates the stream each step (h/h2 are just scratch)
        ops.advance_pos(self.pos.data_ptr(), B, self.smax, stream=s,
                        sync=False)

    def _kv_attn(self, li: int, lay: Dict):
        """Shared per-layer KV append + attention (dense or paged)."""
        ops, s = self._C.ops, self.stream
        B = self.batch
        scale = 1.0 / float(np.sqrt(float(self.hd)))
        if self.kv_pool is not None:
            pool = self.kv_pool
            ops.kv_append_paged(self.qkv.data_ptr(),
                                pool.kpools[li].data_ptr(),
                                pool.vpools[li].data_ptr(),
                                pool.table.data_ptr(),
                                self.pos.data_ptr(), B, self.heads,
                                pool.max_pages, stream=s, sync=False,
                                D=self.hd)
            ops.decode_attention_paged(
                self.qkv.data_ptr(), pool.kpools[li].data_ptr(),
                pool.vpools[li].data_ptr(), self.att.data_ptr(),
                pool.table.data_ptr(), self.pos.data_ptr(), B,
                self.heads, pool.max_pages, scale, stream=s,
                sync=False, D=self.hd)
        else:
            ops.kv_append(self.qkv.data_ptr(), lay["kcache"].data_ptr(),
                          lay["vcache"].data_ptr(), self.pos.data_ptr(),
                          B, self.heads, self.smax, stream=s, sync=False,
                          D=self.hd)
            ops.decode_attention(self.qkv.data_ptr(),
                                 lay["kcache"].data_ptr(),
                                 lay["vcache"].data_ptr(),
                                 self.att.data_ptr(),
                                 self.pos.data_ptr(), B,
                                 self.heads, self.smax, scale,
                                 stream=s, sync=False, D=self.hd)

    def _enqueue_llama(self):
        """One LLaMA decode step (pos-relative, captured like the gpt2
        recipe): embed -> [per layer: rmsnorm -> qkv gemm -> rope(pos) ->
        kv_append -> decode_attention -> proj -> add_rmsnorm(res fused) ->
        gate/up gemms -> silu_mul -> down -> add_rmsnorm] -> rms_f ->
        [lm head] -> advance_pos. No biases anywhere (LLaMA); RoPE reads
        the device position counter directly (pos_dev[row], idle slots
        skip), so the captured graph stays position-free."""
        C, s = self._C, self.stream
        B, Hd = self.batch, self.hidden
        T4 = 0
        ops = C.ops
        ops.decode_embed(self.ids.data_ptr(), self.tok.data_ptr(),
                         self.posemb.data_ptr(), self.h.data_ptr(),
                         self.pos.data_ptr(), B, Hd, stream=s, sync=False)
        ops.rmsnorm(0, self.h.data_ptr(),
                    self.layers[0]["rms1_g"].data_ptr(), self.x.data_ptr(),
                    B, Hd, eps=self.rms_eps, stream=s, sync=False)
        for li, lay in enumerate(self.layers):
            ops.gemm_bt(0, self.x.data_ptr(), lay["qkv_w"].data_ptr(),
                        self.qkv.data_ptr(), M=B, N=3 * Hd, K=Hd,
                        epi=self._epi_none, stream=s, sync=False, tile=T4)
            ops.rope(0, self.qkv.data_ptr(), pos=self.pos.data_ptr(), M=B,
                     S=self.smax, H=self.heads, D=self.hd, theta=self.theta,
                     stream=s, sync=False)
            self._kv_attn(li, lay)
            ops.gemm_bt(0, self.att.data_ptr(), lay["proj_w"].data_ptr(),
                        self.x2.data_ptr(), M=B, N=Hd, K=Hd,
                        epi=self._epi_none, stream=s, sync=False, tile=T4)
            # h += proj; x = rms2(h)  (one kernel, sum_out = new residual)
            ops.add_rmsnorm(0, self.x2.data_ptr(), self.h.data_ptr(),
                            lay["rms2_g"].data_ptr(), self.x.data_ptr(),
                            sum_out=self.h.data_ptr(), M=B, N=Hd,
                            eps=self.rms_eps, stream=s, sync=False)
            ops.gemm_bt(0, self.x.data_ptr(), lay["gate_w"].data_ptr(),
                        self.ff.data_ptr(), M=B, N=self.inter, K=Hd,
                        epi=self._epi_none, stream=s, sync=False, tile=T4)
            ops.gemm_bt(0, self.x.data_ptr(), lay["up_w"].data_ptr(),
                        self.ff2.data_ptr(), M=B, N=self.inter, K=Hd,
                        epi=self._epi_none, stream=s, sync=False, tile=T4)
            ops.silu_mul(0, self.ff.data_ptr(), self.ff2.data_ptr(),
                         self.ff.data_ptr(), B * self.inter, stream=s,
                         sync=False)
            ops.gemm_bt(0, self.ff.data_ptr(), lay["down_w"].data_ptr(),
                        self.x2.data_ptr(), M=B, N=Hd, K=self.inter,
                        epi=self._epi_none, stream=s, sync=False, tile=T4)
            nxt = (self.layers[li + 1] if li + 1 < self.n_layers else None)
            gptr = (nxt["rms1_g"] if nxt else self.lnf_g).data_ptr()
            dst = (self.x if nxt else self.out).data_ptr()
            ops.add_rmsnorm(0, self.x2.data_ptr(), self.h.data_ptr(), gptr,
                            dst, sum_out=self.h.data_ptr(), M=B, N=Hd,
                            eps=self.rms_eps, stream=s, sync=False)
        if self.logits is not None:
            ops.gemm_bt(0, self.out.data_ptr(), self.tok.data_ptr(),
                        self.logits.data_ptr(), M=B, N=self.vocab, K=Hd,
                        epi=self._epi_none, stream=s, sync=False)
            ops.argmax_rows(self.logits.data_ptr(), self.gids.data_ptr(),
                            B, self.vocab, stream=s, sync=False)
        ops.advance_pos(self.pos.data_ptr(), B, self.smax, stream=s,
                        sync=False)

    def _enqueue(self):
        """Record one decode step's kernels on self.stream (pos-relative:
        kv_append/decode_attention/embed all read the device counter).

        The step is kernel-COUNT bound (~4.6 us graph-replay floor per
        kernel regardless of size - profiles/dec_kernel_stats), so every
        residual add is fused into the following layernorm (sum_out
        updates the residual stream in the same kernel) and split-K is
        disabled on the tiny M=B gemms via the tile hint (the fp32-slab
        reduce kernel doubled the gemm count for no win at this floor).
        """
        if getattr(self, "arch", "gpt2") == "llama":
            self._enqueue_llama()
            return
        if getattr(self, "fused", False):
            self._enqueue_fused()
            return
        C, s = self._C, self.stream
        B, Hd = self.batch, self.hidden
        T4 = 0  # heuristic tiles (split-K allowed)
        ops = C.ops
        ops.decode_embed(self.ids.data_ptr(), self.tok.data_ptr(),
                         self.posemb.data_ptr(), self.h.data_ptr(),
                         self.pos.data_ptr(), B, Hd, stream=s, sync=False)
        ops.layernorm(0, self.h.data_ptr(),
                      self.layers[0]["ln1_g"].data_ptr(),
                      self.layers[0]["ln1_b"].data_ptr(), self.x.data_ptr(),
                      B, Hd, stream=s, sync=False)
        for li, lay in enumerate(self.layers):
            ops.gemm_bt(0, self.x.data_ptr(), lay["qkv_w"].data_ptr(),
                        self.qkv.data_ptr(), bias=lay["qkv_b"].data_ptr(),
                        M=B, N=3 * Hd, K=Hd, epi=self._epi_bias, stream=s,
                        sync=False, tile=T4)
            self._kv_attn(li, lay)
            ops.gemm_bt(0, self.att.data_ptr(), lay["proj_w"].data_ptr(),
                        self.x2.data_ptr(), bias=lay["proj_b"].data_ptr(),
                        M=B, N=Hd, K=Hd, epi=self._epi_bias, stream=s,
                        sync=False, tile=T4)
            # h += proj; x = ln2(h)   (one kernel: sum_out = new residual)
            ops.add_layernorm(0, self.x2.data_ptr(), self.h.data_ptr(),
                              lay["ln2_g"].data_ptr(),
                              lay["ln2_b"].data_ptr(), self.x.data_ptr(),
                              sum_out=self.h.data_ptr(), M=B, N=Hd,
                              stream=s, sync=False)
            ops.gemm_bt(0, self.x.data_ptr(), lay["ff1_w"].data_ptr(),
                        self.ff.data_ptr(), bias=lay["ff1_b"].data_ptr(),
                        M=B, N=self.inter, K=Hd, epi=self._epi_gelu,
                        stream=s, sync=False, tile=T4)
            ops.gemm_bt(0, self.ff.data_ptr(), lay["ff2_w"].data_ptr(),
                        self.x2.data_ptr(), bias=lay["ff2_b"].data_ptr(),
                        M=B, N=Hd, K=self.inter, epi=self._epi_bias,
                        stream=s, sync=False, tile=T4)
            # h += ff2; x = next ln1(h) (or ln_f at the end)
            nxt = (self.layers[li + 1] if li + 1 < self.n_layers else None)
            gptr = (nxt["ln1_g"] if nxt else self.lnf_g).data_ptr()
            bptr = (nxt["ln1_b"] if nxt else self.lnf_b).data_ptr()
            dst = (self.x if nxt else self.out).data_ptr()
            ops.add_layernorm(0, self.x2.data_ptr(), self.h.data_ptr(),
                              gptr, bptr, dst, sum_out=self.h.data_ptr(),
                              M=B, N=Hd, stream=s, sync=False)
        if self.logits is not None:
            ops.gemm_bt(0, self.out.data_ptr(), self.tok.data_ptr(),
                        self.logits.data_ptr(), M=B, N=self.vocab, K=Hd,
                        epi=self._epi_none, stream=s, sync=False)
            ops.argmax_rows(self.logits.data_ptr(), self.gids.data_ptr(),
                            B, self.vocab, stream=s, sync=False)
        ops.advance_pos(self.pos.data_ptr(), B, self.smax, stream=s,
                        sync=False)

    def _kv_range(self, li: int, lay: Dict, qkv_ptr: int, B: int,
                  Pp: int) -> None:
        """Prefill cache writer (dense or paged)."""
        ops, s = self._C.ops, self.stream
        if self.kv_pool is not None:
            pool = self.kv_pool
            ops.kv_append_range_paged(qkv_ptr, pool.kpools[li].data_ptr(),
                                      pool.vpools[li].data_ptr(),
                                      pool.table.data_ptr(), B, self.heads,
                                      Pp, pool.max_pages, stream=s,
                                      sync=False, D=self.hd)
        else:
            ops.kv_append_range(qkv_ptr, lay["kcache"].data_ptr(),
                                lay["vcache"].data_ptr(), B, self.heads,
                                Pp, self.smax, stream=s, sync=False,
                                D=self.hd)

    def prefill(self, prompt: np.ndarray) -> np.ndarray:
        """Fill the KV caches from a whole prompt [B, P] in ONE pass through
        the full-sequence kernels (causal online-softmax attention), then
        continue with step() from position P — prefill runs at the
        parallel-forward rate (~2M tok/s) instead of one replay per token.
        P is padded to the next multiple of 128 internally; the causal mask
        keeps positions < P exact and the padded cache slots are
        overwritten before they are ever attended. Returns the final
        hidden state at position P-1 per sequence. Must be called before
        the first step()."""
        import torch

        if self._steps != 0 or self._graph:
            raise RuntimeError("prefill must run before the first step()")
        prompt = np.ascontiguousarray(prompt, np.int32)
        B, P = prompt.shape
        assert B == self.batch and 0 < P < self.smax
        Pp = (P + 127) // 128 * 128
        if Pp > self.posemb.shape[0]:
            raise RuntimeError(
                f"prefill: padded prompt ({Pp}) exceeds the position table "
                f"({int(self.posemb.shape[0])}) — shorten the prompt or "
                "build the model with a longer seq")
        ids = np.zeros((B, Pp), np.int32)
        ids[:, :P] = prompt
        if self.kv_pool is not None:
            # map pages covering the PADDED prompt (the tail rows are
            # written by kv_append_range_paged and overwritten by later
            # decode steps before they are ever attended)
            for b in range(B):
                for li in range((Pp - 1) // 64 + 1):
                    self.kv_pool.ensure(b, li)
        M = B * Pp
        Hd, inter = self.hidden, self.inter
        ops, s = self._C.ops, self.stream

        dids = torch.from_numpy(ids.reshape(-1)).cuda()
        h = torch.empty(M, Hd, dtype=torch.half, device="cuda")
        x = torch.empty(M, Hd, dtype=torch.half, device="cuda")
        x2 = torch.empty(M, Hd, dtype=torch.half, device="cuda")
        qkv = torch.empty(M, 3 * Hd, dtype=torch.half, device="cuda")
        ff = torch.empty(M, inter, dtype=torch.half, device="cuda")
        ff2 = (torch.empty(M, inter, dtype=torch.half, device="cuda")
               if self.arch == "llama" else None)
        torch.cuda.synchronize()

        scale = 1.0 / float(np.sqrt(float(self.hd)))
        ops.embedding(0, dids.data_ptr(), self.tok.data_ptr(),
                      self.posemb.data_ptr(), out=h.data_ptr(), M=M, S=Pp,
                      H=Hd, stream=s, sync=False)
        if self.arch == "llama":
            ops.rmsnorm(0, h.data_ptr(),
                        self.layers[0]["rms1_g"].data_ptr(), x.data_ptr(),
                        M, Hd, eps=self.rms_eps, stream=s, sync=False)
            for li, lay in enumerate(self.layers):
                ops.gemm_bt(0, x.data_ptr(), lay["qkv_w"].data_ptr(),
                            qkv.data_ptr(), M=M, N=3 * Hd, K=Hd,
                            epi=self._epi_none, stream=s, sync=False)
                # full-sequence rope: pos = row % Pp (exact for rows < P;
                # padded tail is causal-masked / overwritten before read)
                ops.rope(0, qkv.data_ptr(), M=M, S=Pp, H=self.heads,
                         D=self.hd, theta=self.theta, stream=s, sync=False)
                self._kv_range(li, lay, qkv.data_ptr(), B, Pp)
                ops.attention(0, qkv.data_ptr(), x2.data_ptr(), B, Pp,
                              self.heads, self.hd, scale, stream=s,
                              sync=False, causal=1)
                ops.gemm_bt(0, x2.data_ptr(), lay["proj_w"].data_ptr(),
                            x.data_ptr(), M=M, N=Hd, K=Hd,
                            epi=self._epi_none, stream=s, sync=False)
                ops.add_rmsnorm(0, x.data_ptr(), h.data_ptr(),
                                lay["rms2_g"].data_ptr(), x2.data_ptr(),
                                sum_out=h.data_ptr(), M=M, N=Hd, eps=self.rms_eps, stream=s,
                                sync=False)
                ops.gemm_bt(0, x2.data_ptr(), lay["gate_w"].data_ptr(),
                            ff.data_ptr(), M=M, N=inter, K=Hd,
                            epi=self._epi_none, stream=s, sync=False)
                ops.gemm_bt(0, x2.data_ptr(), lay["up_w"].data_ptr(),
                            ff2.data_ptr(), M=M, N=inter, K=Hd,
                            epi=self._epi_none, stream=s, sync=False)
                ops.silu_mul(0, ff.data_ptr(), ff2.data_ptr(),
                             ff.data_ptr(), M * inter, stream=s, sync=False)
                ops.gemm_bt(0, ff.data_ptr(), lay["down_w"].data_ptr(),
                            x2.data_ptr(), M=M, N=Hd, K=inter,
                            epi=self._epi_none, stream=s, sync=False)
                nxt = (self.layers[li + 1] if li + 1 < self.n_layers
                       else None)
                gptr = (nxt["rms1_g"] if nxt else self.lnf_g).data_ptr()
                ops.add_rmsnorm(0, x2.data_ptr(), h.data_ptr(), gptr,
                                x.data_ptr(), sum_out=h.data_ptr(), M=M,
                                N=Hd, eps=self.rms_eps, stream=s, sync=False)
        else:
            ops.layernorm(0, h.data_ptr(),
                          self.layers[0]["ln1_g"].data_ptr(),
                          self.layers[0]["ln1_b"].data_ptr(), x.data_ptr(),
                          M, Hd, stream=s, sync=False)
            for li, lay in enumerate(self.layers):
                ops.gemm_bt(0, x.data_ptr(), lay["qkv_w"].data_ptr(),
                            qkv.data_ptr(), bias=lay["qkv_b"].data_ptr(),
                            M=M, N=3 * Hd, K=Hd, epi=self._epi_bias,
                            stream=s, sync=False)
                self._kv_range(li, lay, qkv.data_ptr(), B, Pp)
                ops.attention(0, qkv.data_ptr(), x2.data_ptr(), B, Pp,
                              self.heads, self.hd, scale,
                              stream=s, sync=False, causal=1)
                ops.gemm_bt(0, x2.data_ptr(), lay["proj_w"].data_ptr(),
                            x.data_ptr(), bias=lay["proj_b"].data_ptr(),
                            M=M, N=Hd, K=Hd, epi=self._epi_bias, stream=s,
                            sync=False)
                ops.add_layernorm(0, x.data_ptr(), h.data_ptr(),
                                  lay["ln2_g"].data_ptr(),
                                  lay["ln2_b"].data_ptr(), x2.data_ptr(),
                                  sum_out=h.data_ptr(), M=M, N=Hd, stream=s,
                                  sync=False)
                ops.gemm_bt(0, x2.data_ptr(), lay["ff1_w"].data_ptr(),
                            ff.data_ptr(), bias=lay["ff1_b"].data_ptr(),
                            M=M, N=inter, K=Hd, epi=self._epi_gelu,
                            stream=s, sync=False)
                ops.gemm_bt(0, ff.data_ptr(), lay["ff2_w"].data_ptr(),
                            x2.data_ptr(), bias=lay["ff2_b"].data_ptr(),
                            M=M, N=Hd, K=inter, epi=self._epi_bias,
                            stream=s, sync=False)
                nxt = (self.layers[li + 1] if li + 1 < self.n_layers
                       else None)
                gptr = (nxt["ln1_g"] if nxt else self.lnf_g).data_ptr()
                bptr = (nxt["ln1_b"] if nxt else self.lnf_b).data_ptr()
                ops.add_layernorm(0, x2.data_ptr(), h.data_ptr(), gptr,
                                  bptr, x.data_ptr(),
                                  sum_out=h.data_ptr(), M=M, N=Hd,
                                  stream=s, sync=False)
        self._C.hip.stream_synchronize(s)
        # x holds ln_f(h) for every position; hand back the last real one
        last = x.reshape(B, Pp, Hd)[:, P - 1].contiguous()
        torch.cuda.synchronize()
        self.pos.fill_(P)
        self._active[:] = True
        self._steps = P
        self._slot_steps[:] = P
        torch.cuda.synchronize()
        if self.logits is not None:
            ops.gemm_bt(0, last.data_ptr(), self.tok.data_ptr(),
                        self.logits.data_ptr(), M=B, N=self.vocab, K=Hd,
                        epi=self._epi_none, stream=s, sync=True)
            return self.logits.float().cpu().numpy()
        return last.float().cpu().numpy()

    def step(self, ids: np.ndarray,
             return_ids: bool = False) -> np.ndarray:
        """Feed one token per sequence; returns the final hidden state
        [B, hidden] fp32 (or logits [B, vocab] with lm_head=True). The
        device-side position counter starts at 0 and the captured graph
        advances it, so replays need no host-side position plumbing.

        return_ids=True (lm_head only): return the greedy argmax token
        per slot [B] int32 instead of logits — the argmax runs inside
        the captured graph, so only B ints cross PCIe per step."""
        if return_ids and self.logits is None:
            raise RuntimeError("return_ids requires lm_head=True")
        if self.kv_pool is not None:
            # map the page each active slot's next write lands in
            for b in range(self.batch):
                if self._active[b]:
                    self.kv_pool.ensure(b, int(self._slot_steps[b]) >> 6)
        if (self._slot_steps[self._active] >= self.smax).any():
            raise RuntimeError(
                "DecodeSession: a slot hit the sequence limit "
                "(reset_slot() it or end the session)")
        arr = np.ascontiguousarray(ids, np.int32)
        # synchronous H2D keeps the token copy ordered before the replay
        self._C.memory.memcpy_h2d(self.ids.data_ptr(), arr, arr.nbytes)
        if self.capture:
            if not self._graph:
                # step 0 runs eagerly as the warm-up, then a fresh step is
                # RECORDED (capture does not execute) for replay from step 1
                self._enqueue()
                self._C.hip.stream_synchronize(self.stream)
                self._C.hip.stream_begin_capture(self.stream)
                self._enqueue()
                self._graph = self._C.hip.stream_end_capture(self.stream)
                self._steps += 1
                self._slot_steps += 1
                if return_ids:
                    return self.gids.cpu().numpy()
                out = self.out.float().cpu().numpy()
                return (self.logits.float().cpu().numpy()
                        if self.logits is not None else out)
            self._C.hip.graph_launch(self._graph, self.stream)
        else:
            self._enqueue()
        self._C.hip.stream_synchronize(self.stream)
        self._steps += 1
        self._slot_steps += 1
        if return_ids:
            return self.gids.cpu().numpy()
        out = self.out.float().cpu().numpy()
        return (self.logits.float().cpu().numpy()
                if self.logits is not None else out)

    # ------------------------------------------- speculative verification
    def _kv_attn_chunk(self, li: int, lay: Dict, cb: Dict, B: int,
                       K: int) -> None:
        """Per-layer chunk KV append + multi-query attention (dense or
        paged — the paged path resolves rows through the page table)."""
        ops, s = self._C.ops, self.stream
        scale = 1.0 / float(np.sqrt(float(self.hd)))
        if self.kv_pool is not None:
            pool = self.kv_pool
            ops.kv_append_chunk_paged(cb["qkv"].data_ptr(),
                                      pool.kpools[li].data_ptr(),
                                      pool.vpools[li].data_ptr(),
                                      pool.table.data_ptr(),
                                      self.pos.data_ptr(), B, self.heads,
                                      K, pool.max_pages, stream=s,
                                      sync=False, D=self.hd)
            ops.chunk_attention_paged(cb["qkv"].data_ptr(),
                                      pool.kpools[li].data_ptr(),
                                      pool.vpools[li].data_ptr(),
                                      cb["att"].data_ptr(),
                                      pool.table.data_ptr(),
                                      self.pos.data_ptr(), B, self.heads,
                                      K, pool.max_pages, scale, stream=s,
                                      sync=False, D=self.hd)
        else:
            ops.kv_append_chunk(cb["qkv"].data_ptr(),
                                lay["kcache"].data_ptr(),
                                lay["vcache"].data_ptr(),
                                self.pos.data_ptr(), B, self.heads, K,
                                self.smax, stream=s, sync=False, D=self.hd)
            ops.chunk_attention(cb["qkv"].data_ptr(),
                                lay["kcache"].data_ptr(),
                                lay["vcache"].data_ptr(),
                                cb["att"].data_ptr(), self.pos.data_ptr(),
                                B, self.heads, K, self.smax, scale,
                                stream=s, sync=False, D=self.hd)

    def _enqueue_chunk(self, cb: Dict, B: int, K: int) -> None:
        """Record one chunked-verification pass's kernels on the session
        stream (no syncs — capturable). All position dependence reads the
        device counter, so a captured chunk graph replays at any pos."""
        M = B * K
        Hd, inter = self.hidden, self.inter
        ops, s = self._C.ops, self.stream
        scale = 1.0 / float(np.sqrt(float(self.hd)))
        ops.chunk_embed(cb["ids"].data_ptr(), self.tok.data_ptr(),
                        self.posemb.data_ptr(), cb["h"].data_ptr(),
                        self.pos.data_ptr(), B, K, self.smax, Hd, stream=s,
                        sync=False)
        if self.arch == "llama":
            ops.rmsnorm(0, cb["h"].data_ptr(),
                        self.layers[0]["rms1_g"].data_ptr(),
                        cb["x"].data_ptr(), M, Hd, eps=self.rms_eps, stream=s, sync=False)
            for li, lay in enumerate(self.layers):
                ops.gemm_bt(0, cb["x"].data_ptr(), lay["qkv_w"].data_ptr(),
                            cb["qkv"].data_ptr(), M=M, N=3 * Hd, K=Hd,
                            epi=self._epi_none, stream=s, sync=False)
                # chunk rope: row b*K+i rotates at pos[b] + i
                ops.rope(0, cb["qkv"].data_ptr(), pos=self.pos.data_ptr(),
                         M=M, S=self.smax, H=self.heads, D=self.hd,
                         theta=self.theta, stream=s, sync=False, chunk=K)
                self._kv_attn_chunk(li, lay, cb, B, K)
                ops.gemm_bt(0, cb["att"].data_ptr(),
                            lay["proj_w"].data_ptr(), cb["x2"].data_ptr(),
                            M=M, N=Hd, K=Hd, epi=self._epi_none, stream=s,
                            sync=False)
                ops.add_rmsnorm(0, cb["x2"].data_ptr(), cb["h"].data_ptr(),
                                lay["rms2_g"].data_ptr(),
                                cb["x"].data_ptr(),
                                sum_out=cb["h"].data_ptr(), M=M, N=Hd,
                                eps=self.rms_eps, stream=s, sync=False)
                ops.gemm_bt(0, cb["x"].data_ptr(), lay["gate_w"].data_ptr(),
                            cb["ff"].data_ptr(), M=M, N=inter, K=Hd,
                            epi=self._epi_none, stream=s, sync=False)
                ops.gemm_bt(0, cb["x"].data_ptr(), lay["up_w"].data_ptr(),
                            cb["ff2"].data_ptr(), M=M, N=inter, K=Hd,
                            epi=self._epi_none, stream=s, sync=False)
                ops.silu_mul(0, cb["ff"].data_ptr(), cb["ff2"].data_ptr(),
                             cb["ff"].data_ptr(), M * inter, stream=s,
                             sync=False)
                ops.gemm_bt(0, cb["ff"].data_ptr(),
                            lay["down_w"].data_ptr(), cb["x2"].data_ptr(),
                            M=M, N=Hd, K=inter, epi=self._epi_none,
                            stream=s, sync=False)
                nxt = (self.layers[li + 1] if li + 1 < self.n_layers
                       else None)
                gptr = (nxt["rms1_g"] if nxt else self.lnf_g).data_ptr()
                ops.add_rmsnorm(0, cb["x2"].data_ptr(), cb["h"].data_ptr(),
                                gptr, cb["x"].data_ptr(),
                                sum_out=cb["h"].data_ptr(), M=M, N=Hd,
                                eps=self.rms_eps, stream=s, sync=False)
        else:
            ops.layernorm(0, cb["h"].data_ptr(),
                          self.layers[0]["ln1_g"].data_ptr(),
                          self.layers[0]["ln1_b"].data_ptr(),
                          cb["x"].data_ptr(), M, Hd, stream=s, sync=False)
            for li, lay in enumerate(self.layers):
                ops.gemm_bt(0, cb["x"].data_ptr(), lay["qkv_w"].data_ptr(),
                            cb["qkv"].data_ptr(),
                            bias=lay["qkv_b"].data_ptr(),
                            M=M, N=3 * Hd, K=Hd, epi=self._epi_bias,
                            stream=s, sync=False)
                self._kv_attn_chunk(li, lay, cb, B, K)
                ops.gemm_bt(0, cb["att"].data_ptr(),
                            lay["proj_w"].data_ptr(), cb["x2"].data_ptr(),
                            bias=lay["proj_b"].data_ptr(),
                            M=M, N=Hd, K=Hd, epi=self._epi_bias, stream=s,
                            sync=False)
                ops.add_layernorm(0, cb["x2"].data_ptr(),
                                  cb["h"].data_ptr(),
                                  lay["ln2_g"].data_ptr(),
                                  lay["ln2_b"].data_ptr(),
                                  cb["x"].data_ptr(),
                                  sum_out=cb["h"].data_ptr(), M=M, N=Hd,
                                  stream=s, sync=False)
                ops.gemm_bt(0, cb["x"].data_ptr(), lay["ff1_w"].data_ptr(),
                            cb["ff"].data_ptr(),
                            bias=lay["ff1_b"].data_ptr(),
                            M=M, N=inter, K=Hd, epi=self._epi_gelu,
                            stream=s, sync=False)
                ops.gemm_bt(0, cb["ff"].data_ptr(), lay["ff2_w"].data_ptr(),
                            cb["x2"].data_ptr(),
                            bias=lay["ff2_b"].data_ptr(),
                            M=M, N=Hd, K=inter, epi=self._epi_bias,
                            stream=s, sync=False)
                nxt = (self.layers[li + 1] if li + 1 < self.n_layers
                       else None)
                gptr = (nxt["ln1_g"] if nxt else self.lnf_g).data_ptr()
                bptr = (nxt["ln1_b"] if nxt else self.lnf_b).data_ptr()
                ops.add_layernorm(0, cb["x2"].data_ptr(),
                                  cb["h"].data_ptr(),
                                  gptr, bptr, cb["x"].data_ptr(),
                                  sum_out=cb["h"].data_ptr(), M=M, N=Hd,
                                  stream=s, sync=False)
        ops.gemm_bt(0, cb["x"].data_ptr(), self.tok.data_ptr(),
                    cb["logits"].data_ptr(), M=M, N=self.vocab, K=Hd,
                    epi=self._epi_none, stream=s, sync=False)
        ops.argmax_rows(cb["logits"].data_ptr(), cb["gids"].data_ptr(),
                        M, self.vocab, stream=s, sync=False)

    def verify_chunk(self, tokens: np.ndarray,
                     greedy: bool = False) -> np.ndarray:
        """Score K proposed tokens per slot in ONE chunked forward
        (speculative decoding's verification step): tokens [B, K] are
        consumed at positions pos[b]..pos[b]+K-1 (their K/V overwrite any
        stale entries), and the TARGET logits for each position come back
        as [B, K, vocab] fp32. pos is NOT advanced — call add_pos() with
        the per-slot accepted counts. Requires lm_head=True.

        With capture=True the pass is hipGraph-captured PER CHUNK SIZE on
        first use and replayed afterwards (the measured spec-decode cost
        was eager per-kernel launch overhead; all position dependence is
        device-side, so one graph serves every position)."""
        import torch

        if self.logits is None:
            raise RuntimeError("verify_chunk requires lm_head=True")
        tokens = np.ascontiguousarray(tokens, np.int32)
        B, K = tokens.shape
        if self.kv_pool is not None:
            # map every page the chunk's writes land in (pos..pos+K-1)
            for b in range(self.batch):
                if self._active[b]:
                    p0 = int(self._slot_steps[b])
                    for li in range(p0 >> 6, ((p0 + K - 1) >> 6) + 1):
                        self.kv_pool.ensure(b, li)
        assert B == self.batch and K >= 1
        M = B * K
        Hd, inter = self.hidden, self.inter
        s = self.stream

        cbs = getattr(self, "_chunk_bufs_by_k", None)
        if cbs is None:
            cbs = self._chunk_bufs_by_k = {}
            self._chunk_graphs = {}
        cb = cbs.get(K)
        if cb is None:
            cb = cbs[K] = dict(
                K=K,
                ids=torch.zeros(M, dtype=torch.int32, device="cuda"),
                h=torch.zeros(M, Hd, dtype=torch.half, device="cuda"),
                x=torch.zeros(M, Hd, dtype=torch.half, device="cuda"),
                x2=torch.zeros(M, Hd, dtype=torch.half, device="cuda"),
                qkv=torch.zeros(M, 3 * Hd, dtype=torch.half, device="cuda"),
                att=torch.zeros(M, Hd, dtype=torch.half, device="cuda"),
                ff=torch.zeros(M, inter, dtype=torch.half, device="cuda"),
                ff2=(torch.zeros(M, inter, dtype=torch.half, device="cuda")
                     if self.arch == "llama" else None),
                logits=torch.zeros(M, self.vocab, dtype=torch.half,
                                   device="cuda"),
                gids=torch.zeros(M, dtype=torch.int32, device="cuda"),
            )
        self._C.memory.memcpy_h2d(cb["ids"].data_ptr(), tokens.reshape(-1),
                                  tokens.nbytes)
        if self.capture:
            gph = self._chunk_graphs.get(K)
            if gph is None:
                # first call: eager warm-up does the real work, then a
                # fresh pass is RECORDED (capture does not execute)
                self._enqueue_chunk(cb, B, K)
                self._C.hip.stream_synchronize(s)
                self._C.hip.stream_begin_capture(s)
                self._enqueue_chunk(cb, B, K)
                self._chunk_graphs[K] = self._C.hip.stream_end_capture(s)
            else:
                self._C.hip.graph_launch(gph, s)
                self._C.hip.stream_synchronize(s)
        else:
            self._enqueue_chunk(cb, B, K)
            self._C.hip.stream_synchronize(s)
        if greedy:
            return cb["gids"].cpu().numpy().reshape(B, K)
        return cb["logits"].float().cpu().numpy().reshape(B, K, self.vocab)

    def sample_tokens(self, temps: np.ndarray,
                      seeds: np.ndarray) -> np.ndarray:
        """Device-side temperature sampling of the LAST step's logits
        (Gumbel-max: an exact softmax(logits/T) categorical draw per
        slot; temps[b] <= 0 degrades to greedy argmax). Noise is keyed
        (seeds[b], device pos[b], index), so consecutive steps draw
        fresh noise and a (seed, position) pair replays identically.
        Only B ints cross PCIe."""
        if self.logits is None:
            raise RuntimeError("sample_tokens requires lm_head=True")
        torch = self._torch
        if getattr(self, "_temps_dev", None) is None:
            self._temps_dev = torch.zeros(self.batch, dtype=torch.float32,
                                          device="cuda")
            self._seeds_dev = torch.zeros(self.batch, dtype=torch.int32,
                                          device="cuda")
        self._C.memory.memcpy_h2d(self._temps_dev.data_ptr(),
                                  np.ascontiguousarray(temps, np.float32),
                                  self.batch * 4)
        self._C.memory.memcpy_h2d(self._seeds_dev.data_ptr(),
                                  np.ascontiguousarray(seeds, np.int32),
                                  self.batch * 4)
        self._C.ops.gumbel_argmax_rows(
            self.logits.data_ptr(), self.gids.data_ptr(),
            temps=self._temps_dev.data_ptr(),
            seeds=self._seeds_dev.data_ptr(), pos=self.pos.data_ptr(),
            M=self.batch, V=self.vocab, stream=self.stream, sync=True)
        return self.gids.cpu().numpy()

    def get_pos(self) -> np.ndarray:
        """Per-slot positions from the HOST mirror (no D2H round-trip:
        device pos only changes via step()'s advance kernel — mirrored
        by _slot_steps — or via these host-side setters)."""
        return np.where(self._active, self._slot_steps, -1).astype(np.int32)

    def add_pos(self, counts: np.ndarray) -> None:
        """Advance each slot's position by counts[b] (speculative
        acceptance: the chunk's first counts[b] tokens are now consumed)."""
        counts = np.asarray(counts, np.int64)
        cur = self.get_pos().astype(np.int64)
        new = np.where(cur >= 0, np.minimum(cur + counts, self.smax - 1),
                       cur)
        self.pos.copy_(self._torch.from_numpy(new.astype(np.int32)).cuda())
        self._slot_steps = np.where(cur >= 0,
                                    self._slot_steps + counts,
                                    self._slot_steps)
        self._torch.cuda.synchronize()

    def set_pos(self, positions: np.ndarray) -> None:
        """Set per-slot absolute positions (speculative rollback of a
        draft session to the target's accepted frontier)."""
        arr = np.ascontiguousarray(positions, np.int32)
        self.pos.copy_(self._torch.from_numpy(arr).cuda())
        self._slot_steps = arr.astype(np.int64).clip(min=0)
        self._active = arr >= 0
        self._torch.cuda.synchronize()

    def reset_slot(self, b: int) -> None:
        """Restart slot b at position 0 (continuous batching in lockstep:
        the next step() token for this slot begins a fresh sequence while
        the other slots keep decoding against their caches)."""
        self.pos[b] = 0
        self._slot_steps[b] = 0
        self._active[b] = True
        if self.kv_pool is not None:
            self.kv_pool.free_slot(b)  # fresh sequence: recycle its pages
        self._torch.cuda.synchronize()

    def idle_slot(self, b: int) -> None:
        """Park slot b (no active request): pos[b] = -1 makes every
        per-slot decode kernel early-exit for it, so an empty slot in the
        continuous batch costs ~nothing per replayed step. reset_slot(b)
        re-activates it for a fresh sequence."""
        self.pos[b] = -1
        self._slot_steps[b] = 0
        self._active[b] = False
        if self.kv_pool is not None:
            self.kv_pool.free_slot(b)  # parked slots hold ZERO cache pages
        self._torch.cuda.synchronize()

    def close(self):
        if self._graph:
            self._C.hip.graph_destroy(self._graph)
            self._graph = 0
        for g in getattr(self, "_chunk_graphs", {}).values():
            self._C.hip.graph_destroy(g)
        if getattr(self, "_chunk_graphs", None):
            self._chunk_graphs = {}

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class SpeculativeDecoder:
    """Greedy speculative decoding over two DecodeSessions (beyond-
    reference serving capability): a cheap DRAFT model proposes k tokens
    per round, the TARGET verifies all k in ONE chunked forward
    (verify_chunk), and the longest matching prefix is accepted plus the
    target's own correction token. The emitted stream is IDENTICAL to
    greedy decoding with the target alone — the draft only changes speed,
    never output (the invariant the GPU test checks with a mismatched
    draft). Per-slot positions let different slots accept different
    lengths each round.

    Both sessions need lm_head=True and the same vocab; prime both on the
    same context before generate().
    """

    def __init__(self, target: "DecodeSession", draft: "DecodeSession",
                 k: int = 4):
        if target.logits is None or draft.logits is None:
            raise ValueError("both sessions need lm_head=True")
        if target.vocab != draft.vocab or target.batch != draft.batch:
            raise ValueError("vocab/batch mismatch between target and draft")
        self.target = target
        self.draft = draft
        self.k = k
        self.proposed = 0
        self.accepted = 0

    def generate(self, seed: np.ndarray, steps: int):
        """Greedy-generate `steps` tokens per slot after consuming `seed`
        [B]. Returns (tokens [B, steps] int32, acceptance_rate)."""
        B, k = self.target.batch, self.k
        seed = np.ascontiguousarray(seed, np.int32)
        # ids-only verification (greedy=True): argmax happens on-device,
        # so only token ids ever cross PCIe (B*k ints vs B*k*vocab floats)
        gt = self.target.verify_chunk(seed[:, None], greedy=True)[:, 0]
        self.target.add_pos(np.ones(B, np.int64))
        gd = self.draft.verify_chunk(seed[:, None], greedy=True)[:, 0]
        self.draft.add_pos(np.ones(B, np.int64))

        out = [[] for _ in range(B)]
        while min(len(o) for o in out) < steps:
            # ---- draft chain: k greedy proposals ----
            props = np.zeros((B, k), np.int32)
            cur = gd.astype(np.int32)
            for i in range(k):
                props[:, i] = cur
                gd = self.draft.verify_chunk(cur[:, None],
                                             greedy=True)[:, 0]
                self.draft.add_pos(np.ones(B, np.int64))
                cur = gd.astype(np.int32)
            # ---- target verifies the whole chunk at once ----
            tg = self.target.verify_chunk(props, greedy=True)  # [B, k]
            accept = np.zeros(B, np.int64)
            corr = np.zeros(B, np.int32)
            for b in range(B):
                g = int(gt[b])
                a = 0
                while a < k and props[b, a] == g:
                    out[b].append(g)
                    g = int(tg[b, a])
                    a += 1
                accept[b] = a
                corr[b] = g          # target's own next token
                out[b].append(g)
            self.proposed += B * k
            self.accepted += int(accept.sum())
            # ---- advance target past the accepted prefix, consume the
            # correction token (its K/V overwrites the rejected slot) ----
            self.target.add_pos(accept)
            gt = self.target.verify_chunk(corr[:, None], greedy=True)[:, 0]
            self.target.add_pos(np.ones(B, np.int64))
            # ---- roll the draft back to the target's frontier ----
            self.draft.set_pos(self.target.get_pos() - 1)
            gd = self.draft.verify_chunk(corr[:, None], greedy=True)[:, 0]
            self.draft.add_pos(np.ones(B, np.int64))
        rate = self.accepted / max(self.proposed, 1)
        return (np.array([o[:steps] for o in out], np.int32), rate)
