"""Hand-written bf16 MFMA training path for the wide MLP config
(BASELINE #5: obs=4096, 4x4096 tanh hidden, A=256, bf16).

Routes every GEMM of the wide config — rollout forwards, update-step
forwards, dgrad chain, dW — through the in-tree gfx950 kernels
(ops/hip/bf16_gemm.hip) instead of torch autocast/rocBLAS, with the
epilogues (bias+tanh, dtanh, bias column-sums) fused into the producing
kernels.  Loss math stays the reference formulation: the per-row loss
gradients come from gauss_gh_wide (exact PPO.py:29-40 analytics, bf16
output feeding the dgrad GEMMs directly).

Layout notes (see bf16_gemm.hip header): torch's [out][in] weight layout
IS the MFMA B-operand layout, so forwards need no weight transposes;
dgrad consumes W^T bf16 copies; dW consumes transposed deltas and
activations produced by the fused transpose+colsum kernel (the bias
gradients ride along with the delta transposes).
"""

from __future__ import annotations

import os
from typing import List

import torch


def _round_up(x: int, m: int) -> int:
    return (x + m - 1) // m * m


class WideBF16Path:
    """Per-engine helper owning the bf16 buffers and the update pipeline."""

    def __init__(self, engine):
        from . import require_hip_ext

        self.ext = require_hip_ext()
        self.eng = engine
        self.cfg = engine.cfg
        self.dev = engine.device
        self.D = engine.obs_space.shape[0]
        self.H: List[int] = list(self.cfg.HIDDEN_SIZES)
        self.A = engine.act_space.shape[0]
        self.P = 2 * self.A
        self.KP = _round_up(self.P + 1, 64)  # gh row padding (mm256 K)
        self._weights_dirty = True
        self._f = torch.empty(0, device=self.dev)
        self._b = torch.empty(0, device=self.dev, dtype=torch.bfloat16)
        # bf16 weight copies (refreshed when params change)
        self.W_bf = [torch.empty(h_out, h_in, device=self.dev,
                                 dtype=torch.bfloat16)
                     for h_in, h_out in zip([self.D] + self.H[:-1], self.H)]
        self.Wt_bf = [torch.empty(w.shape[1], w.shape[0], device=self.dev,
                                  dtype=torch.bfloat16) for w in self.W_bf]
        HL = self.H[-1]
        self.whcat = torch.empty(self.P + 1, HL, device=self.dev,
                                 dtype=torch.bfloat16)
        self.whcatT = torch.zeros(HL, self.KP, device=self.dev,
                                  dtype=torch.bfloat16)  # zero K padding
        self.bhcat = torch.empty(self.P + 1, device=self.dev)
        self._fwd_bufs = {}
        self._upd = None

    # -- weights -------------------------------------------------------
    def mark_dirty(self) -> None:
        self._weights_dirty = True

    @torch.no_grad()
    def refresh_weights(self) -> None:
        pi = self.eng.pi
        for l, lay in enumerate(pi.hidden):
            self.W_bf[l].copy_(lay.weight.detach())
            self.ext.bf16_transpose(
                self.W_bf[l], self.Wt_bf[l], self._f, 0,
                self.W_bf[l].shape[0], self.W_bf[l].shape[1],
                self.W_bf[l].shape[1], self.W_bf[l].shape[0])
        self.whcat[:self.P].copy_(pi.pi.weight.detach())
        self.whcat[self.P:].copy_(pi.vf.weight.detach())
        self.ext.bf16_transpose(self.whcat, self.whcatT, self._f, 0,
                                self.P + 1, self.H[-1], self.H[-1], self.KP)
        self.bhcat[:self.P].copy_(pi.pi.bias.detach())
        self.bhcat[self.P:].copy_(pi.vf.bias.detach())
        self._weights_dirty = False

    # -- forward (rollout acting / bootstrap) --------------------------
    def _fwd_scratch(self, M: int):
        key = M
        bufs = self._fwd_bufs.get(key)
        if bufs is None:
            bufs = {
                "x": torch.empty(M, self.D, device=self.dev,
                                 dtype=torch.bfloat16),
                "h": [torch.empty(M, h, device=self.dev,
                                  dtype=torch.bfloat16) for h in self.H],
                "pd": torch.empty(M, self.P, device=self.dev,
                                  dtype=torch.bfloat16),
                "v": torch.empty(M, device=self.dev, dtype=torch.bfloat16),
            }
            self._fwd_bufs[key] = bufs
        return bufs

    @staticmethod
    def _dualw() -> bool:
        """Transposed dual-writes from the GEMM epilogues (default on;
        DPPO_WIDE_DUALW=0 restores the separate transpose passes)."""
        return os.environ.get("DPPO_WIDE_DUALW", "1") != "0"

    @torch.no_grad()
    def forward(self, obs: torch.Tensor, out=None, ct=None):
        """(v, pdflat) in f32, every GEMM on the hand bf16 kernels.

        `out = (h_views, pd_view, v_view)` redirects the layer outputs
        into caller-owned buffers — the graphed rollout records its
        activations straight into the update pipeline's buffers so the
        first update step can skip its forward entirely (the parameters
        are unchanged between rollout and step 1).  `ct = (hT_views,
        ldt)` additionally dual-writes the TRANSPOSED activations
        (column-offset views of the update pipeline's hT buffers), so
        step 1 skips its hT transposes too."""
        if self._weights_dirty:
            self.refresh_weights()
        M = obs.shape[0]
        s = self._fwd_scratch(M)
        if out is None:
            hs, pd_o, v_o = s["h"], s["pd"], s["v"]
        else:
            hs, pd_o, v_o = out
        if out is not None:  # recording pass: remember whether hT rode along
            self._hT_from_rollout = ct is not None
        s["x"].copy_(obs)
        x = s["x"]
        pi = self.eng.pi
        for l in range(len(self.H)):
            if ct is not None:
                self.ext.bf16_mm256(x, self.W_bf[l], hs[l], 1,
                                    pi.hidden[l].bias.detach(), self._b,
                                    self._f, 0, ct[0][l], ct[1],
                                    self._f, 0)
            else:
                self.ext.bf16_mm256(x, self.W_bf[l], hs[l], 1,
                                    pi.hidden[l].bias.detach(), self._b,
                                    self._f, 0, self._b, 0, self._f, 0)
            x = hs[l]
        self.ext.bf16_mm_small(x, self.whcat, pd_o, v_o, self._b,
                               self._f, 0, 0, 0, 4, M, self.P + 1, self.P,
                               self.bhcat)
        return v_o.float(), pd_o.float()

    # -- update --------------------------------------------------------
    def _upd_bufs(self, B: int):
        if self._upd is not None and self._upd["B"] == B:
            return self._upd
        dev = self.dev
        bf = torch.bfloat16
        u = {
            "B": B,
            "x": torch.empty(B, self.D, device=dev, dtype=bf),
            "xT": torch.empty(self.D, B, device=dev, dtype=bf),
            "h": [torch.empty(B, h, device=dev, dtype=bf) for h in self.H],
            # transposed activations, dual-written by the forward GEMMs
            "hT": [torch.empty(h, B, device=dev, dtype=bf) for h in self.H],
            "d0": torch.empty(B, max(self.H), device=dev, dtype=bf),
            "d1": torch.empty(B, max(self.H), device=dev, dtype=bf),
            "dT0": torch.empty(max(self.H), B, device=dev, dtype=bf),
            "dT1": torch.empty(max(self.H), B, device=dev, dtype=bf),
            "pd": torch.empty(B, self.P, device=dev, dtype=bf),
            "v": torch.empty(B, device=dev, dtype=bf),
            "gh": torch.zeros(B, self.KP, device=dev, dtype=bf),
            "ghT": torch.empty(self.KP, B, device=dev, dtype=bf),
            "bias_tmp": torch.zeros(self.KP, device=dev),
            # persistent input copies: the captured update graph needs
            # stable addresses (the eager rollout reallocates per round)
            "oldflat": torch.empty(B, self.P, device=dev),
            "oldv": torch.empty(B, device=dev),
            "actions": torch.empty(B, self.A, device=dev),
            "adv": torch.empty(B, device=dev),
            "etr": torch.empty(B, device=dev),
            "clip_dev": torch.zeros(1, device=dev),
        }
        self._upd = u
        self._graph = None
        return u

    @torch.no_grad()
    def update(self, batch, l_mul: float) -> None:
        """UPDATE_STEPS repeated full-batch steps (Chief.py:64), every
        GEMM hand-written: fwd (tanh+bias fused) -> gh -> dgrad chain
        (dtanh fused) -> transposes (+bias colsums fused) -> dW into the
        f32 flat grad -> all-reduce -> fused Adam."""
        eng, ext, cfg = self.eng, self.ext, self.cfg
        B = batch.states.shape[0]
        u = self._upd_bufs(B)
        clip = cfg.CLIP_PARAM * l_mul
        nH = len(self.H)
        flat = eng.flat_pi
        offsets = [sl.start for sl in flat.slices]
        w_off = [offsets[2 * l] for l in range(nH)]
        b_off = [offsets[2 * l + 1] for l in range(nH)]
        off_wv, off_bv = offsets[2 * nH], offsets[2 * nH + 1]
        off_wp, off_bp = offsets[2 * nH + 2], offsets[2 * nH + 3]
        grad = flat.flat_grad

        # per-round inputs (constant across the UPDATE_STEPS epochs),
        # copied into persistent buffers so the captured graph sees them
        u["x"].copy_(batch.states)
        ext.bf16_transpose(u["x"], u["xT"], self._f, 0, B, self.D, self.D, B)
        u["oldflat"].copy_(batch.oldflat)
        u["oldv"].copy_(batch.oldv)
        u["actions"].copy_(batch.actions)
        u["adv"].copy_(batch.adv)
        u["etr"].copy_(batch.etr)
        u["clip_dev"].fill_(clip)
        eng.optimizer.lr_dev.fill_(float(eng.optimizer.param_groups[0]["lr"]))

        import os as _os
        # hipGraph capture of a multi-stream (event-forked) body segfaults
        # in capture_end on this ROCm build, so the captured variant runs
        # the SEQUENTIAL body (measured fastest: the side-stream overlap
        # bought nothing — the 1-block/CU GEMMs leave no residency for
        # co-scheduled transpose blocks).  DPPO_WIDE_GRAPH=0 opts out.
        graph_ok = (cfg.USE_GRAPHS
                    and _os.environ.get("DPPO_WIDE_GRAPH") != "0"
                    and (not eng.comm.distributed
                         or _os.environ.get("DPPO_GRAPH_DIST") == "1")
                    and not getattr(self, "_graph_failed", False))
        skip_now = bool(getattr(eng, "_wide_rollout_h_valid", False))
        if graph_ok:
            if getattr(self, "_graph_skip_first", None) is not None and \
                    self._graph_skip_first != skip_now:
                self._graph = None  # recapture with the new structure
            if self._graph is None:
                try:
                    self._capture(u)
                except Exception as exc:  # noqa: BLE001
                    eng._warn_once(
                        "wide_graph",
                        f"wide-update hipGraph capture failed ({exc!r}); "
                        "running uncaptured")
                    self._graph_failed = True
                    self._graph = None
            if self._graph is not None:
                self._graph.replay()
                self._weights_dirty = True
                return
        self._update_body(u)

    def _capture(self, u) -> None:
        """Capture the UPDATE_STEPS pipeline once (device-state Adam,
        clip from clip_dev; warmup executes real steps so param/optimizer
        state is snapshot/restored around it)."""
        eng = self.eng
        opt = eng.optimizer
        snap = (eng.flat_pi.flat_param.detach().clone(), opt.exp_avg.clone(),
                opt.exp_avg_sq.clone(), opt.step_dev.clone())
        try:
            warm = torch.cuda.Stream()
            warm.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(warm):
                self._update_body(u, overlap=False)
            torch.cuda.current_stream().wait_stream(warm)
            torch.cuda.synchronize()
        finally:
            with torch.no_grad():
                eng.flat_pi.flat_param.copy_(snap[0])
                opt.exp_avg.copy_(snap[1])
                opt.exp_avg_sq.copy_(snap[2])
                opt.step_dev.copy_(snap[3])
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._update_body(u, overlap=False)
        self._graph = g
        self._graph_skip_first = bool(
            getattr(self.eng, "_wide_rollout_h_valid", False))

    @torch.no_grad()
    def _update_body(self, u, overlap: bool = False) -> None:
        """UPDATE_STEPS pipeline with the memory-bound transposes and
        heads-dW overlapped on a side stream: the 256^2 GEMMs run 2
        waves/SIMD MFMA-bound (6 idle wave slots per SIMD and <1/3 of
        HBM bandwidth used), so the 4.4 TB/s transpose kernel
        co-schedules nearly for free.  Cross-stream edges are events
        (graph capture records them)."""
        eng, ext, cfg = self.eng, self.ext, self.cfg
        B = u["B"]
        nH = len(self.H)
        flat = eng.flat_pi
        offsets = [sl.start for sl in flat.slices]
        w_off = [offsets[2 * l] for l in range(nH)]
        b_off = [offsets[2 * l + 1] for l in range(nH)]
        off_wv, off_bv = offsets[2 * nH], offsets[2 * nH + 1]
        off_wp, off_bp = offsets[2 * nH + 2], offsets[2 * nH + 3]
        grad = flat.flat_grad
        main = torch.cuda.current_stream()
        if getattr(self, "_side", None) is None:
            self._side = torch.cuda.Stream()
        side = self._side if overlap else main
        evs = self._evs = []  # keep alive across graph replays

        class _NullCtx:
            def __enter__(self):
                return None

            def __exit__(self, *a):
                return False

        side_ctx = (lambda: torch.cuda.stream(side)) if overlap \
            else (lambda: _NullCtx())

        def fork():
            if not overlap:
                return
            e = torch.cuda.Event()
            evs.append(e)
            e.record(main)
            side.wait_event(e)

        def join():
            if not overlap:
                return None
            e = torch.cuda.Event()
            evs.append(e)
            e.record(side)
            return e

        # When the graphed rollout recorded its activations into u["h"]/
        # u["pd"]/u["v"] (same parameters as step 1 — sync_oldpi then
        # rollout then update, no param change in between), step 1 skips
        # its forward+heads GEMMs entirely.
        skip_first = bool(getattr(eng, "_wide_rollout_h_valid", False))
        # transposed-operand dual-writes from the GEMM epilogues (the
        # LDS-repacked stripes store 32 contiguous bytes per column) in
        # place of 8 of the 9 separate transpose passes per step:
        # measured +3.9% same-box (580.5 vs 558.9 K env-steps/s);
        # DPPO_WIDE_DUALW=0 restores the separate transpose passes
        dualw = os.environ.get("DPPO_WIDE_DUALW", "1") != "0"
        for step_i in range(cfg.UPDATE_STEPS):
            self.refresh_weights()
            x = u["x"]
            ev_hT = [None] * nH
            if step_i == 0 and skip_first:
                if not (dualw and getattr(self, "_hT_from_rollout", False)):
                    for l in range(nH):
                        fork()
                        with side_ctx():
                            ext.bf16_transpose(u["h"][l], u["hT"][l],
                                               self._f, 0, B, self.H[l],
                                               self.H[l], B)
                        ev_hT[l] = join()
            else:
                for l in range(nH):
                    if dualw:
                        ext.bf16_mm256(x, self.W_bf[l], u["h"][l], 1,
                                       eng.pi.hidden[l].bias.detach(),
                                       self._b, self._f, 0,
                                       u["hT"][l], B, self._f, 0)
                    else:
                        ext.bf16_mm256(x, self.W_bf[l], u["h"][l], 1,
                                       eng.pi.hidden[l].bias.detach(),
                                       self._b, self._f, 0, self._b, 0,
                                       self._f, 0)
                        fork()
                        with side_ctx():
                            ext.bf16_transpose(u["h"][l], u["hT"][l],
                                               self._f, 0, B, self.H[l],
                                               self.H[l], B)
                        ev_hT[l] = join()
                    x = u["h"][l]
                ext.bf16_mm_small(x, self.whcat, u["pd"], u["v"], self._b,
                                  self._f, 0, 0, 0, 4, B, self.P + 1,
                                  self.P, self.bhcat)
            # loss gradients -> gh [B][P+1] (bf16, K-padded)
            ext.gauss_gh_wide(u["pd"], u["oldflat"], u["v"], u["oldv"],
                              u["actions"], u["adv"], u["etr"], u["gh"],
                              u["clip_dev"], 0.0, cfg.ENTCOEFF, cfg.VCOEFF)
            flat.zero_grad()
            # side: ghT (+ head bias colsums) and the whole heads dW —
            # independent of the main dgrad/dW chain
            fork()
            with side_ctx():
                u["bias_tmp"].zero_()
                ext.bf16_transpose(u["gh"], u["ghT"], u["bias_tmp"], 0,
                                   B, self.KP, self.KP, B)
                grad[off_bp:off_bp + self.P].copy_(u["bias_tmp"][:self.P])
                grad[off_bv:off_bv + 1].copy_(
                    u["bias_tmp"][self.P:self.P + 1])
                if overlap:
                    side.wait_event(ev_hT[nH - 1])
                ext.bf16_mm_small(u["ghT"], u["hT"][nH - 1], self._b,
                                  self._b, self._b, grad, off_wp, off_wv,
                                  self.P, 5, self.P + 1, self.H[-1], 0,
                                  self._f)
            ev_heads = join()
            # heads backward: dz[nH-1] = (gh @ Whcat) * dtanh(h[-1])
            d_cur = u["d0"]
            dT_cur = u["dT0"]
            if dualw:
                ext.bf16_mm256(u["gh"], self.whcatT, d_cur, 2, self._f,
                               u["h"][nH - 1], self._f, 0,
                               dT_cur, B, grad, b_off[nH - 1])
            else:
                ext.bf16_mm256(u["gh"], self.whcatT, d_cur, 2, self._f,
                               u["h"][nH - 1], self._f, 0,
                               self._b, 0, self._f, 0)
                fork()
                with side_ctx():
                    ext.bf16_transpose(d_cur, dT_cur, grad, b_off[nH - 1],
                                       B, self.H[nH - 1], self.H[nH - 1], B)
                ev_dT = join()
            # heads gradients are the LAST contiguous flat-grad region
            # (vf.w | vf.b | pi.w | pi.b): all-reduce them as their own
            # bucket as soon as they are complete, overlapping the rest
            # of the backward (the wide config's ~276 MB f32
            # grads are bandwidth-bound — bucket-per-layer overlap; the
            # tiny flagship grads keep the single fused bucket)
            per_layer_ar = eng.comm.distributed
            if per_layer_ar:
                if overlap:
                    main.wait_event(ev_heads)
                eng.comm.allreduce_mean_(grad[off_wv:])
            # hidden chain (uniform H: buffers are exact-size views)
            for l in range(nH - 1, -1, -1):
                actT = u["xT"] if l == 0 else u["hT"][l - 1]
                if overlap:
                    if l > 0:
                        main.wait_event(ev_hT[l - 1])
                    main.wait_event(ev_dT)
                ext.bf16_mm256(dT_cur, actT, self._b, 3, self._f, self._b,
                               grad, w_off[l], self._b, 0, self._f, 0)
                if per_layer_ar and l > 0:
                    # W_l and b_l are adjacent in the flat layout; b_l's
                    # colsum landed with dz_l's transpose, dW_l just now
                    eng.comm.allreduce_mean_(
                        grad[w_off[l]:b_off[l] + self.H[l]])
                if l > 0:
                    # dz[l-1] = (dz[l] @ W[l]) * dtanh(h[l-1]); side
                    # transposes dz[l-1] (+ colsum -> db[l-1])
                    d_nxt = u["d1"] if d_cur is u["d0"] else u["d0"]
                    dT_nxt = u["dT1"] if dT_cur is u["dT0"] else u["dT0"]
                    if dualw:
                        ext.bf16_mm256(d_cur, self.Wt_bf[l], d_nxt, 2,
                                       self._f, u["h"][l - 1], self._f, 0,
                                       dT_nxt, B, grad, b_off[l - 1])
                    else:
                        ext.bf16_mm256(d_cur, self.Wt_bf[l], d_nxt, 2,
                                       self._f, u["h"][l - 1], self._f, 0,
                                       self._b, 0, self._f, 0)
                        fork()
                        with side_ctx():
                            ext.bf16_transpose(d_nxt, dT_nxt, grad,
                                               b_off[l - 1], B,
                                               self.H[l - 1],
                                               self.H[l - 1], B)
                        ev_dT = join()
                    d_cur, dT_cur = d_nxt, dT_nxt
            if overlap and not per_layer_ar:
                main.wait_event(ev_heads)
            if per_layer_ar:
                # last bucket: layer 0's W/b (its bias colsum rides dz_0's
                # transpose which precedes dW_0 in stream order)
                eng.comm.allreduce_mean_(grad[w_off[0]:b_off[0] + self.H[0]])
            else:
                eng.comm.allreduce_mean_(grad)
            eng.optimizer.step_captured()  # lr_dev set by the caller
            self._weights_dirty = True
