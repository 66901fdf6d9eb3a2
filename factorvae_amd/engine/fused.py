"""Fused MI355X training engine.

Runs the whole FactorVAE training step — forward, hand-written backward,
flat-bucket gradient all-reduce, fused Adam + device-side cosine LR —
through the hand-written HIP kernels in factorvae_amd/ops/hip/, with the
kernel sequence captured into a hipGraph (one replay per trading-day
step; the reference pays ~100 eager ATen/cuDNN launches plus a D2H sync
per step, /root/reference/train_model.py:26-32).

Design:
- every parameter lives in ONE flat fp32 arena ordered so that the K
  attention heads' {query, key W, key b, value W, value b} are contiguous
  stacked (K,H,H)/(K,H) views (zero-copy kernel inputs; the module's
  Parameters are rebound to arena views so state_dict()/load_state_dict
  keep the reference checkpoint contract);
- gradients live in a mirror arena: backward kernels write/accumulate
  into views, the DP all-reduce is one RCCL call on the flat buffer, and
  Adam is one kernel over the arena;
- eps / dropout-mask RNG uses torch's generator OUTSIDE the graph (two
  tiny kernels per step), keeping set_seed reproducibility;
- four HIP streams per step: the main activation chain, two side
  streams for the weight-gradient reductions, and (in DP runs) a comm
  stream whose early all-reduce of the non-extractor gradient slice
  overlaps the extractor backward;
- loss stays on device; reading it is the caller's (async) choice.
"""

from __future__ import annotations

import math
import os
import warnings
from typing import Dict, Optional

import torch

from ..models.modules import FactorVAE
from ..ops import get_extension
from ..parallel.ddp import get_world_size, is_distributed


class UnsupportedShapeError(ValueError):
    """Model shape outside the fused kernels' tiling envelope; callers
    fall back to the eager (PyTorch-ROCm) engine."""


class _Arena:
    """Flat fp32 buffer + named views with a custom packing order."""

    def __init__(self, device, specs):
        # specs: list of (name, shape)
        self.offsets = {}
        total = 0
        for name, shape in specs:
            n = 1
            for s in shape:
                n *= s
            self.offsets[name] = (total, shape)
            total += n
        self.flat = torch.zeros(total, device=device, dtype=torch.float32)

    def view(self, name):
        off, shape = self.offsets[name]
        n = 1
        for s in shape:
            n *= s
        return self.flat[off:off + n].view(*shape)


class FusedTrainer:
    """Fused training engine for one FactorVAE model on one GPU rank."""

    DROPOUT_P = 0.1

    def __init__(self, model: FactorVAE, lr: float, t_max: int,
                 device: Optional[torch.device] = None, eta_min: float = 0.0,
                 use_graph: bool = True, max_stocks: Optional[int] = None,
                 train: bool = True, dtype: str = "fp32"):
        assert dtype in ("fp32", "bf16", "fp8")
        self.ext = get_extension()
        # bf16 mode (BASELINE.json configs 2-4): the FLOP-bound extractor
        # GEMM family (R = N*T rows) runs on bf16 MFMA (~2.5 PF/s dense on
        # gfx950 vs 157 TF/s f32) with fp32 accumulate; fp32 master
        # weights + fp32 Adam + fp32 grad all-reduce; the latency-bound
        # N-row kernels (encoder/attention/decoder/GRU recurrence) stay
        # fp32 — their cost is dispatch+LDS, not FLOPs.
        # fp8 mode (config 5): forward extractor GEMMs on fp8 e4m3 MFMA
        # (weights+activations; per-tensor weight scales, LN-normalized
        # activations at unit scale); backward reuses the bf16 path.
        self.bf16 = dtype in ("bf16", "fp8")
        self.fp8 = dtype == "fp8"
        self.dtype = dtype
        self.model = model
        self.device = device or torch.device("cuda")
        model.to(self.device)
        self.lr = lr
        self.eta_min = eta_min
        self.t_max = t_max
        self.training = train
        # FV_GRAPH=0: eager-launch every kernel (no hipGraph capture at
        # all) — the last-resort guard rail for capture-hostile boxes
        self.use_graph = use_graph and os.environ.get("FV_GRAPH",
                                                       "1") != "0"

        fe = model.feature_extractor
        self.C = fe.num_latent
        self.H = fe.hidden_size
        enc = model.factor_encoder
        self.M = enc.linear.out_features
        self.K = model.factor_predictor.num_factor
        if self.H > 64:
            # the GRU/attention kernel register/LDS tilings are sized for
            # H <= 64 (all reference checkpoints: H in {20, 32, 64});
            # callers catch this and fall back to the eager engine
            raise UnsupportedShapeError(
                f"fused engine supports hidden_size <= 64, got {self.H}; "
                f"use engine='eager'")

        self._build_param_arena()
        # bf16/fp8 @ H=64: Whh wgrad computed inside the GRU backward
        # kernel (FV_WHH_FUSED=0 restores the standalone TN call)
        self._whh_fused = (self.bf16 and self.H == 64
                           and os.environ.get("FV_WHH_FUSED", "1") != "0")
        if self.bf16:
            C3 = 3 * self.H
            # padded (+ transposed-padded) weight shadows for the
            # register-stationary NT kernel: k dim zero-padded to a
            # multiple of 32 so the MFMA k-tail is exact (pads are
            # zeroed here once and never written by the refresh kernel)
            KPc = (self.C + 31) & ~31
            KP3 = (C3 + 31) & ~31
            zb = lambda *s: torch.zeros(*s, dtype=torch.bfloat16,
                                        device=self.device)
            self.w1x_p = zb(self.C, KPc)    # W1x (C,C) k-padded
            self.w1xT_p = zb(self.C, KPc)   # W1x^T
            self.wih_p = zb(C3, KPc)        # Wih (3H,C) k-padded
            self.wihT_p = zb(self.C, KP3)   # Wih^T (C,3H) k-padded
            self.whh_bf = torch.empty(C3, self.H, dtype=torch.bfloat16,
                                      device=self.device)
            if self.fp8:
                # row stride padded to 128 for the MX-scaled K=128 MFMA
                # path (the only fp8 form at the ~5 PF/s rate); pads are
                # zeroed here and never written (casts write Ci cols)
                ldp = ((self.C + 127) & ~127)
                self._fp8_rs = ldp <= 256
                if not self._fp8_rs:  # fallback: K=32 fp8 MFMA path
                    ldp = (self.C + 3) & ~3
                f8 = torch.float8_e4m3fn
                self.w1x_f8 = torch.zeros(self.C, ldp, dtype=f8,
                                          device=self.device)
                self.wih_f8 = torch.zeros(C3, ldp, dtype=f8,
                                          device=self.device)
                self.s_w1x = torch.ones(1, device=self.device)
                self.is_w1x = torch.ones(1, device=self.device)
                self.s_wih = torch.ones(1, device=self.device)
                self.is_wih = torch.ones(1, device=self.device)
                if self._fp8_rs:
                    # fp8 dgrad path: transposed padded weight shadows +
                    # delayed-scaling state for the two dgrad operands
                    # (amax collected this step -> scale next step)
                    KPt3 = (C3 + 127) & ~127
                    KPtC = (self.C + 127) & ~127
                    self.wihT_f8 = torch.zeros(self.C, KPt3, dtype=f8,
                                               device=self.device)
                    self.w1xT_f8 = torch.zeros(self.C, KPtC, dtype=f8,
                                               device=self.device)
                    one = lambda v: torch.full((1,), float(v),
                                               device=self.device)
                    self.amax_dgi = one(1.0)
                    self.s_dgi = one(448.0)
                    self.is_dgi = one(1.0 / 448.0)
                    self.amax_dzx = one(1.0)
                    self.s_dzx = one(448.0)
                    self.is_dzx = one(1.0 / 448.0)
            self._refresh_bf16_shadows()
        self.grads = torch.zeros_like(self.params.flat)
        self.adam_m = torch.zeros_like(self.params.flat)
        self.adam_v = torch.zeros_like(self.params.flat)
        self.step_t = torch.zeros(1, device=self.device, dtype=torch.int32)

        self._ws_n = 0
        self._ws_cache: Dict = {}
        self._graphs: Dict = {}
        self._g_inputs = None
        # side stream: weight-gradient GEMMs/colsums run here, overlapped
        # with the activation-gradient critical path (fork/join via events;
        # the dependencies are recorded into the captured hipGraph)
        self.s_side = (torch.cuda.Stream(device=self.device)
                       if self.device.type == "cuda" else None)
        # second side stream: the three big extractor wgrad reductions
        # run here so they overlap BOTH the main path and the other
        # (attention/encoder) wgrads — at A-share shapes the side work
        # exceeds the main path, so one side stream becomes the bottleneck
        self.s_side2 = (torch.cuda.Stream(device=self.device)
                        if self.device.type == "cuda" else None)
        # FV_MAIN_PRIO=1 (opt-in, default OFF): run the critical path on
        # a HIGH-priority stream. Measured MUCH slower on gfx950
        # (CSI300 2804 -> 1338 cs/s; A-share 591 -> 531): replaying the
        # graph from a priority queue appears to serialize the captured
        # side branches. Kept as a documented negative-result knob.
        self.s_main = None
        if (self.device.type == "cuda"
                and os.environ.get("FV_MAIN_PRIO", "0") == "1"):
            self.s_main = torch.cuda.Stream(device=self.device, priority=-1)
        # comm stream: in DP runs the attention/encoder/decoder slice of
        # the gradient arena (everything packed after the extractor) is
        # all-reduced here as soon as its last producer finishes,
        # overlapped with the extractor backward; the extractor slice
        # follows after the join (xGMI ring latency hides under ~40% of
        # the backward)
        self.s_comm = (torch.cuda.Stream(device=self.device)
                       if self.device.type == "cuda" else None)

    # ---------------------------------------------------------------- params
    def _param_specs(self):
        m = self.model
        C, H, M, K = self.C, self.H, self.M, self.K
        specs = []

        def add(name, param):
            specs.append((name, param, tuple(param.shape)))

        fe = m.feature_extractor
        add("ln_g", fe.normalize.weight)
        add("ln_b", fe.normalize.bias)
        add("W1x", fe.linear.weight)
        add("b1x", fe.linear.bias)
        add("Wih", fe.gru.weight_ih_l0)
        add("Whh", fe.gru.weight_hh_l0)
        add("bih", fe.gru.bias_ih_l0)
        add("bhh", fe.gru.bias_hh_l0)

        enc = m.factor_encoder
        add("Wenc", enc.linear.weight)
        add("benc", enc.linear.bias)
        add("Wmu_e", enc.linear_mu.weight)
        add("bmu_e", enc.linear_mu.bias)
        add("Wsig_e", enc.linear_sigma.weight)
        add("bsig_e", enc.linear_sigma.bias)

        dec = m.factor_decoder
        add("W1d", dec.alpha_layer.linear1.weight)
        add("b1d", dec.alpha_layer.linear1.bias)
        add("wmu_d", dec.alpha_layer.mu_layer.weight)
        add("bmu_d", dec.alpha_layer.mu_layer.bias)
        add("wsig_d", dec.alpha_layer.sigma_layer.weight)
        add("bsig_d", dec.alpha_layer.sigma_layer.bias)
        add("Wb", dec.beta_layer.linear1.weight)
        add("bb", dec.beta_layer.linear1.bias)

        pred = m.factor_predictor
        # stacked attention groups: heads consecutive -> contiguous (K,..)
        for i, layer in enumerate(pred.attention_layers):
            add(f"q_att.{i}", layer.query)
        for i, layer in enumerate(pred.attention_layers):
            add(f"Wk.{i}", layer.key_layer.weight)
        for i, layer in enumerate(pred.attention_layers):
            add(f"bk.{i}", layer.key_layer.bias)
        for i, layer in enumerate(pred.attention_layers):
            add(f"Wv.{i}", layer.value_layer.weight)
        for i, layer in enumerate(pred.attention_layers):
            add(f"bv.{i}", layer.value_layer.bias)

        add("Wl", pred.linear.weight)
        add("bl", pred.linear.bias)
        add("wmu_p", pred.mu_layer.weight)
        add("bmu_p", pred.mu_layer.bias)
        add("wsig_p", pred.sigma_layer.weight)
        add("bsig_p", pred.sigma_layer.bias)
        return specs

    def _build_param_arena(self):
        specs = self._param_specs()
        self.params = _Arena(self.device, [(n, s) for n, _, s in specs])
        for name, param, _ in specs:
            v = self.params.view(name)
            v.copy_(param.data)
            param.data = v
        # stacked cross-head views
        K, H = self.K, self.H
        off0, _ = self.params.offsets["q_att.0"]
        self.p_q = self.params.flat[off0:off0 + K * H].view(K, H)
        offW, _ = self.params.offsets["Wk.0"]
        self.p_Wk = self.params.flat[offW:offW + K * H * H].view(K, H, H)
        offb, _ = self.params.offsets["bk.0"]
        self.p_bk = self.params.flat[offb:offb + K * H].view(K, H)
        offV, _ = self.params.offsets["Wv.0"]
        self.p_Wv = self.params.flat[offV:offV + K * H * H].view(K, H, H)
        offvb, _ = self.params.offsets["bv.0"]
        self.p_bv = self.params.flat[offvb:offvb + K * H].view(K, H)

    def p(self, name):
        return self.params.view(name)

    def g(self, name):
        off, shape = self.params.offsets[name]
        n = 1
        for s in shape:
            n *= s
        return self.grads[off:off + n].view(*shape)

    def _gstack(self, base, shape):
        off, _ = self.params.offsets[base]
        n = 1
        for s in shape:
            n *= s
        return self.grads[off:off + n].view(*shape)

    # ------------------------------------------------------------ workspaces
    def _alloc_ws(self, N: int, T: int):
        d = self.device
        C, H, M, K = self.C, self.H, self.M, self.K
        R = N * T
        f = lambda *shape: torch.zeros(*shape, device=d, dtype=torch.float32)
        w = {}
        w["x"] = f(N, T, C)
        w["y"] = f(N, 1)
        w["xln"] = f(R, C)
        w["mean"] = f(R)
        w["rstd"] = f(R)
        w["xp"] = f(R, C)
        w["gi"] = f(R, 3 * H)
        w["h"] = f(N, H)
        w["h_prev"] = f(N, T, H)
        w["gates4"] = f(N, T, 4 * H)
        w["scores_enc"] = f(N, M)
        w["a_enc"] = f(N, M)
        w["yp"] = f(M)
        w["fmu"] = f(K)
        w["fsig_pre"] = f(K)
        w["fsig"] = f(K)
        w["fsig_c"] = f(K)
        w["qk"] = f(K, H)
        w["cb"] = f(K)
        w["s_att"] = f(N, K)
        w["mask"] = f(N, K)
        w["a_att"] = f(N, K)
        w["sd"] = f(N, K)
        w["guard"] = torch.zeros(K, device=d, dtype=torch.int32)
        w["enc_done"] = torch.zeros(1, device=d, dtype=torch.int32)
        w["u"] = f(K, H)
        w["ctx"] = f(K, H)
        w["hm2"] = f(K, H)
        w["pmu"] = f(K)
        w["psig_pre"] = f(K)
        w["psig"] = f(K)
        w["psig_c"] = f(K)
        w["a1"] = f(N, H)
        w["beta"] = f(N, K)
        w["asig_pre"] = f(N)
        w["sigma"] = f(N)
        w["eps"] = f(N)
        w["recon"] = f(N)
        w["loss"] = f(1)
        w["mse"] = f(1)
        w["kl"] = f(1)
        # backward
        w["drecon"] = f(N)
        w["dfmu"] = f(K)
        w["dfsig_c"] = f(K)
        w["dpmu"] = f(K)
        w["dpsig_c"] = f(K)
        w["dh"] = f(N, H)
        w["dz1"] = f(N, H)
        w["dbeta"] = f(N, K)
        w["dz2"] = f(K, H)
        w["dctx"] = f(K, H)
        w["du"] = f(K, H)
        w["da"] = f(N, K)
        w["ds"] = f(N, K)
        w["dc"] = f(K)
        w["dqk"] = f(K, H)
        w["dscores"] = f(N, M)
        w["dgi"] = f(R, 3 * H)
        w["dgh"] = f(R, 3 * H)
        w["dxp"] = f(R, C)
        max_mn = max(3 * H * max(H, C), C * C, M * H, K * max(H, M))
        w["tn_part"] = f(128 * max_mn)
        w["tn_part2"] = f(128 * max_mn)
        w["tn_part3"] = f(128 * max_mn)
        max_m = max(3 * H, C, M, K)
        w["tn_partb"] = f(128 * max_m)
        w["tn_partb2"] = f(128 * max_m)
        w["tn_partb3"] = f(128 * max_m)
        # N-row reductions (attention/encoder/decoder wgrads + fwd u):
        # chunked over z so a 3500-stock day doesn't serialize 6 blocks
        small_mn = 32 * max(M, K, H) * H
        w["tn_part_u"] = f(small_mn)
        w["tn_part_s"] = f(small_mn)
        w["tn_partb_s"] = f(32 * max_m)
        # attention-branch partials: that branch runs on its own stream
        # concurrently with the dec/enc wgrads that use tn_part_s
        w["tn_part_a"] = f(small_mn)
        # deterministic shared-grad partials (no float atomics anywhere)
        dec_iters = min(8, max(1, (N + 511) // 512))
        dec_nblk = (N + 4 * dec_iters - 1) // (4 * dec_iters) + 1
        w["dec_part"] = f(dec_nblk * (2 * K + 2 * H + 2))
        w["hpart"] = f(K * (2 * H + 2))
        cb = (C + 63) // 64
        ty = (2048 + cb - 1) // cb
        # mirror the launcher's rpb clamp to [64, 8192]
        # (extractor.hip fv_ln_bwd_params): without the upper clamp,
        # yblocks at very large R exceeds the partial buffer
        rpb = min(8192, max(64, (R + ty - 1) // ty))
        w["ln_part"] = f(((R + rpb - 1) // rpb + 1) * 2 * C)
        w["dzx"] = f(R, C)
        w["dxln"] = f(R, C)
        if self.bf16:
            fb = lambda *shape: torch.zeros(*shape, device=d,
                                            dtype=torch.bfloat16)
            # A-side operands of the register-stationary NT kernel get
            # 8 elements of tail slack (its k-tail b128 fragment may
            # read up to 4 bytes past the last row; the zero-padded
            # weight makes the contribution exact)
            fbs = lambda r, c: torch.zeros(r * c + 8, device=d,
                                           dtype=torch.bfloat16
                                           )[:r * c].view(r, c)
            w["xln_bf"] = fbs(R, C)
            w["xp_bf"] = fbs(R, C)
            w["dgi_bf"] = fbs(R, 3 * H)
            w["dzx_bf"] = fbs(R, C)
            if self._whh_fused:
                # the in-GRU Whh wgrad replaces the dgh_bf operand image
                # + h_prev cast with per-block partials
                nblk = (N + 15) // 16
                w["whh_part"] = f(nblk * 3 * H * H)
                w["bhh_part"] = f(nblk * 3 * H)
            else:
                w["h_prev_bf"] = fb(R, H)
                w["dgh_bf"] = fb(R, 3 * H)
            if self.fp8:
                ldp = self.w1x_f8.size(1)
                f8t = lambda *shape: torch.zeros(
                    *shape, device=d, dtype=torch.float8_e4m3fn)
                w["xln_f8"] = f8t(R, ldp)
                w["xp_f8"] = f8t(R, ldp)
                if self._fp8_rs:
                    w["dgi_f8"] = f8t(R, self.wihT_f8.size(1))
                    w["dzx_f8"] = f8t(R, self.w1xT_f8.size(1))
        self._ws_cache[(N, T)] = w
        self.ws = w
        self._ws_n = N
        self._ws_t = T

    def _main_ctx(self):
        """Context manager entering the high-priority main stream (or a
        no-op when unavailable/disabled)."""
        if self.s_main is not None:
            return torch.cuda.stream(self.s_main)
        import contextlib
        return contextlib.nullcontext()

    def _main_entry(self):
        """Order s_main after work the caller enqueued on its stream
        (input tensors)."""
        if self.s_main is not None:
            self.s_main.wait_stream(torch.cuda.current_stream(self.device))

    def _main_exit(self):
        """Order the caller's stream after the step's s_main work (the
        caller reads the device loss/scores)."""
        if self.s_main is not None:
            torch.cuda.current_stream(self.device).wait_stream(self.s_main)

    # ------------------------------------------------------ stream helpers
    def _streams(self):
        return (self.s_side, self.s_side2)

    def _fork(self, si: int = 0):
        """Make side stream si wait for everything issued on main."""
        sides = self._streams()
        if sides[si] is None:
            return
        e = torch.cuda.Event()
        e.record(torch.cuda.current_stream(self.device))
        sides[si].wait_event(e)

    def _join(self, si: int = 0):
        """Make main wait for everything issued on side stream si."""
        sides = self._streams()
        if sides[si] is None:
            return
        e = torch.cuda.Event()
        e.record(sides[si])
        torch.cuda.current_stream(self.device).wait_event(e)

    class _OnSide:
        """Context manager: run enqueues on a side stream (no-op when the
        stream is unavailable, e.g. CPU)."""

        def __init__(self, trainer, si: int = 0):
            self.stream = trainer._streams()[si]

        def __enter__(self):
            if self.stream is not None:
                self.ctx = torch.cuda.stream(self.stream)
                self.ctx.__enter__()
            return self

        def __exit__(self, *a):
            if self.stream is not None:
                self.ctx.__exit__(*a)
            return False

    # ------------------------------------------------------------ the step
    def _launch_forward(self, N: int, T: int, with_loss: bool = True,
                        x=None, y=None):
        ext, w, p = self.ext, self.ws, self.p
        C, H, M, K = self.C, self.H, self.M, self.K
        R = N * T
        x2d = (w["x"] if x is None else x).view(R, C)
        yv = w["y"] if y is None else y
        alpha = 1.0 / math.sqrt(float(H) + 1e-6)

        # the K-head query/key projection depends only on parameters:
        # issue it on the side stream overlapped with the extractor
        self._fork(0)
        with self._OnSide(self, 0):
            ext.attn_qk_fwd(self.p_q, self.p_Wk, self.p_bk, w["qk"], w["cb"])

        if self.fp8:
            # fwd in e4m3 (weights+activations); also emits the bf16
            # activation copies the bf16 backward consumes
            ext.ln_fwd(x2d, p("ln_g"), p("ln_b"), None, w["mean"],
                       w["rstd"], 1e-5, w["xln_bf"], w["xln_f8"])
            if self._fp8_rs:
                # MX-scaled K=128 MFMA (double the K=32 form's rate)
                ext.gemm_nt_fp8_rs(w["xln_f8"], self.w1x_f8, p("b1x"),
                                   self.is_w1x, None, w["xp_bf"],
                                   w["xp_f8"], R, self.C, self.C, 1.0, True)
                ext.gemm_nt_fp8_rs(w["xp_f8"], self.wih_f8, p("bih"),
                                   self.is_wih, w["gi"].view(R, 3 * H),
                                   None, None, R, self.C, 3 * H, 1.0, False)
            else:
                ext.gemm_nt_fp8(w["xln_f8"], self.w1x_f8, p("b1x"),
                                self.is_w1x, None, w["xp_bf"], w["xp_f8"],
                                R, self.C, self.C, 1.0, True)
                ext.gemm_nt_fp8(w["xp_f8"], self.wih_f8, p("bih"),
                                self.is_wih, w["gi"].view(R, 3 * H), None,
                                None, R, self.C, 3 * H, 1.0, False)
        elif self.bf16:
            ext.ln_fwd(x2d, p("ln_g"), p("ln_b"), None, w["mean"],
                       w["rstd"], 1e-5, w["xln_bf"])
            ext.gemm_nt_bf16_rs(w["xln_bf"], self.w1x_p, p("b1x"), None,
                                w["xp_bf"], None, 1.0, True)
            ext.gemm_nt_bf16_rs(w["xp_bf"], self.wih_p, p("bih"), w["gi"],
                                None, None, 1.0, False)
        else:
            ext.ln_fwd(x2d, p("ln_g"), p("ln_b"), w["xln"],
                       w["mean"], w["rstd"], 1e-5)
            ext.gemm_nt(w["xln"], p("W1x"), p("b1x"), w["xp"], 1.0, False, True)
            ext.gemm_nt(w["xp"], p("Wih"), p("bih"), w["gi"], 1.0, False, False)
        if self.bf16 and H == 64:
            # h_seq is a dead output in the engine (backward consumes
            # h_prev + gates4); skip its (N,T,H) write entirely
            ext.gru_fwd_mfma(w["gi"], self.whh_bf, p("bhh"), w["h"],
                             None, w["h_prev"], w["gates4"], N, T, H)
        else:
            ext.gru_fwd(w["gi"], p("Whh"), p("bhh"), w["h"], None,
                        w["h_prev"], w["gates4"], N, T, H)
        if self.bf16 and not self._whh_fused:
            # h_prev bf16 image (Whh TN wgrad operand) on side stream 2
            if self.s_side2 is not None:
                e_ = torch.cuda.Event()
                e_.record(torch.cuda.current_stream(self.device))
                self.s_side2.wait_event(e_)
                with torch.cuda.stream(self.s_side2):
                    ext.cast_f32_bf16(w["h_prev"].view(-1),
                                      w["h_prev_bf"].view(-1))
            else:
                ext.cast_f32_bf16(w["h_prev"].view(-1),
                                  w["h_prev_bf"].view(-1))
        # h is ready: the K-head attention branch (prior path) runs on the
        # side stream, overlapped with the encoder + decoder branches on
        # main — they are independent until the loss joins pmu/psig with
        # the posterior-decoded recon
        mask = w["mask"] if self.training else None
        keep_inv = 1.0 / (1.0 - self.DROPOUT_P)
        self._fork(0)
        with self._OnSide(self, 0):
            if N <= 448:
                # whole per-head chain in ONE kernel (h staged in LDS)
                ext.attn_fused_fwd(w["h"], w["qk"], w["cb"], mask, self.p_Wv,
                                   self.p_bv, p("Wl"), p("bl"), p("wmu_p"),
                                   p("bmu_p"), p("wsig_p"), p("bsig_p"),
                                   w["a_att"], w["sd"], w["guard"], w["u"],
                                   w["ctx"], w["hm2"], w["pmu"],
                                   w["psig_pre"], w["psig"], w["psig_c"],
                                   alpha, keep_inv)
            else:
                ext.gemm_nt(w["h"], w["qk"], w["cb"], w["s_att"], alpha,
                            False, False)
                ext.attn_softmax_fwd(w["s_att"], mask, w["a_att"], w["sd"],
                                     w["guard"], keep_inv)
                ext.gemm_tn(w["a_att"], w["h"], w["u"], w["tn_part_u"], 2,
                            False)
                ext.attn_ctx_fwd(w["u"], self.p_Wv, self.p_bv, w["guard"],
                                 w["ctx"])
                ext.pred_mlp_fwd(w["ctx"], p("Wl"), p("bl"), p("wmu_p"),
                                 p("bmu_p"), p("wsig_p"), p("bsig_p"),
                                 w["hm2"], w["pmu"], w["psig_pre"], w["psig"],
                                 w["psig_c"])
        if N <= 448:
            # the mu/sigma heads ride the portfolio kernel (its last
            # workgroup computes them after an agent-scope yp handoff)
            ext.enc_fused_fwd(w["h"], p("Wenc"), p("benc"), yv,
                              w["scores_enc"], w["a_enc"], w["yp"],
                              p("Wmu_e"), p("bmu_e"), p("Wsig_e"),
                              p("bsig_e"), w["fmu"], w["fsig_pre"],
                              w["fsig"], w["fsig_c"], w["enc_done"])
        else:
            ext.gemm_nt(w["h"], p("Wenc"), p("benc"), w["scores_enc"], 1.0,
                        False, False)
            ext.enc_softmax_fwd(w["scores_enc"], yv, w["a_enc"], w["yp"])
            ext.enc_heads_fwd(w["yp"], p("Wmu_e"), p("bmu_e"), p("Wsig_e"),
                              p("bsig_e"), w["fmu"], w["fsig_pre"],
                              w["fsig"], w["fsig_c"])
        ext.dec_fwd(w["h"], p("W1d"), p("b1d"), p("wmu_d"), p("bmu_d"),
                    p("wsig_d"), p("bsig_d"), p("Wb"), p("bb"), w["fmu"],
                    w["fsig_c"], w["eps"], w["recon"], w["a1"], w["beta"],
                    w["asig_pre"], w["sigma"])
        self._join(0)
        if self.bf16:
            # the h_prev bf16 cast was forked onto side stream 2: it must
            # join before a forward-only hipGraph capture ends (an
            # unjoined captured branch invalidates the capture; in the
            # training step the backward's joins covered it)
            self._join(1)
        if with_loss:
            # one kernel: loss scalars AND the five loss input-gradients
            # (the gradient writes cost nothing extra; validation simply
            # ignores them)
            ext.loss_fused(w["recon"], yv, w["fmu"], w["fsig_c"], w["pmu"],
                           w["psig_c"], w["loss"], w["mse"], w["kl"],
                           w["drecon"], w["dfmu"], w["dfsig_c"], w["dpmu"],
                           w["dpsig_c"], 1.0)

    def _launch_backward(self, N: int, T: int, x=None, y=None,
                         comm_overlap: bool = False):
        """Backward pass. Three independent gradient branches run
        concurrently: the decoder+encoder chain on the main stream, the
        K-head attention backward on side stream 2, and every
        weight-gradient GEMM / colsum forked onto side stream 1 as soon
        as its producer is done (fork/join events become hipGraph edges
        under capture). The loss input-gradients were already produced
        by forward's loss_fused kernel; the three dh contributions are
        assembled by ONE dh_combine kernel at the join."""
        ext, w, p, g = self.ext, self.ws, self.p, self.g
        C, H, M, K = self.C, self.H, self.M, self.K
        R = N * T
        x2d = (w["x"] if x is None else x).view(R, C)
        yv = w["y"] if y is None else y
        alpha = 1.0 / math.sqrt(float(H) + 1e-6)
        keep_inv = 1.0 / (1.0 - self.DROPOUT_P)
        chunks = (int(os.environ.get("FV_TN_CHUNKS", "0"))
                  or max(1, min(32, R // 1024)))

        main = torch.cuda.current_stream(self.device) if self.s_side else None
        fork, _on_side = self._fork, self._OnSide
        sides = self._streams()

        # ---- attention backward branch (side stream 2): independent of
        # the decoder/encoder chain given the loss grads from forward
        mask = w["mask"] if self.training else None
        gWv = self._gstack("Wv.0", (K, H, H))
        gbv = self._gstack("bv.0", (K, H))
        gq = self._gstack("q_att.0", (K, H))
        gWk = self._gstack("Wk.0", (K, H, H))
        gbk = self._gstack("bk.0", (K, H))
        fork(1)
        with _on_side(self, 1):
            if N <= 384:
                # whole per-head backward chain in ONE kernel (incl. the
                # query/key/value wgrads)
                ext.attn_fused_bwd(w["dpmu"], w["dpsig_c"], w["psig"],
                                   w["psig_pre"], w["hm2"], p("wmu_p"),
                                   p("wsig_p"), p("Wl"), w["h"], w["a_att"],
                                   w["sd"], mask, w["guard"], w["u"],
                                   self.p_Wv, self.p_q, self.p_Wk, self.p_bk,
                                   w["dz2"], w["du"], w["ds"], w["dc"], gWv,
                                   gbv, gq, gWk, gbk, w["hpart"], g("wmu_p"),
                                   g("bmu_p"), g("wsig_p"), g("bsig_p"),
                                   alpha, keep_inv)
            else:
                ext.pred_mlp_bwd(w["dpmu"], w["dpsig_c"], w["psig"],
                                 w["psig_pre"], w["hm2"], p("wmu_p"),
                                 p("wsig_p"), w["dz2"], w["hpart"],
                                 g("wmu_p"), g("bmu_p"), g("wsig_p"),
                                 g("bsig_p"))
                ext.gemm_nn(w["dz2"], p("Wl"), None, w["dctx"], 1.0, False,
                            False)
                ext.attn_head_bwd(w["dctx"], w["u"], self.p_Wv, w["guard"],
                                  w["du"], gWv, gbv)
                ext.gemm_nt(w["h"], w["du"], None, w["da"], 1.0, False,
                            False)
                ext.attn_softmax_bwd(w["da"], w["a_att"], w["sd"], mask,
                                     w["guard"], w["ds"], w["dc"], keep_inv,
                                     alpha)
                ext.gemm_tn(w["ds"], w["h"], w["dqk"], w["tn_part_a"], 2,
                            False)
                ext.attn_qk_bwd(w["dqk"], w["dc"], self.p_q, self.p_Wk,
                                self.p_bk, gq, gWk, gbk)
            # cross-head Wl wgrad (dz2 is final in both branches)
            ext.gemm_tn(w["dz2"], w["ctx"], g("Wl"), None, 1, False, g("bl"))

        # ---- decoder backward (main)
        ext.dec_bwd(w["drecon"], w["h"], w["a1"], w["beta"], w["asig_pre"],
                    w["sigma"], w["eps"], w["fmu"], w["fsig_c"], p("W1d"),
                    p("wmu_d"), p("wsig_d"), p("Wb"), w["dh"], w["dz1"],
                    w["dbeta"], w["dec_part"], w["dfmu"], w["dfsig_c"],
                    g("wmu_d"), g("bmu_d"), g("wsig_d"), g("bsig_d"))
        fork()
        with _on_side(self):
            ext.gemm_tn(w["dz1"], w["h"], g("W1d"), w["tn_part_s"], 2, False,
                        g("b1d"), w["tn_partb_s"])
            ext.gemm_tn(w["dbeta"], w["h"], g("Wb"), w["tn_part_s"], 2, False,
                        g("bb"), w["tn_partb_s"])

        # ---- encoder backward (heads + stock-axis softmax bwd, main)
        ext.enc_bwd_fused(w["dfmu"], w["dfsig_c"], w["fsig"], w["fsig_pre"],
                          w["yp"], p("Wmu_e"), p("Wsig_e"), w["a_enc"], yv,
                          w["dscores"], g("Wmu_e"), g("bmu_e"), g("Wsig_e"),
                          g("bsig_e"))
        fork()
        with _on_side(self):
            ext.gemm_tn(w["dscores"], w["h"], g("Wenc"), w["tn_part_s"], 2,
                        False, g("benc"), w["tn_partb_s"])

        # ---- join attention branch; assemble all three dh contributions
        # (attention ds@qk + a@du, encoder dscores@Wenc) in ONE kernel
        self._join(1)
        ext.dh_combine(w["dh"], w["ds"], w["qk"], w["a_att"], w["du"],
                       w["dscores"], p("Wenc"))

        if comm_overlap:
            # every non-extractor gradient (arena tail from Wenc on) is
            # final: reduce it now, overlapped with the extractor bwd
            e1 = torch.cuda.Event()
            e1.record(main)
            self.s_comm.wait_event(e1)
            if sides[0] is not None:
                e2 = torch.cuda.Event()
                e2.record(sides[0])
                self.s_comm.wait_event(e2)
            with torch.cuda.stream(self.s_comm):
                tail = self.grads[self.params.offsets["Wenc"][0]:]
                tail.div_(get_world_size())
                torch.distributed.all_reduce(tail)

        # extractor backward
        fp8_rs = self.fp8 and getattr(self, "_fp8_rs", False)
        fp8_gru_fused = fp8_rs and H == 64
        if fp8_rs:
            # turn last step's collected amaxes into this step's operand
            # scales (and reset); must precede the producers below
            ext.scale_from_amax2(self.amax_dgi, self.s_dgi, self.is_dgi,
                                 self.amax_dzx, self.s_dzx, self.is_dzx)
        if self.bf16 and H == 64:
            # fp32 dgi/dgh images are dead in bf16 mode (the wgrads use
            # the bf16 copies); in fp8 mode the kernel also emits the
            # scaled e4m3 dgrad operand directly from registers; with
            # _whh_fused the Whh/bhh wgrad partials come out of the same
            # kernel (no dgh_bf image, no standalone TN call)
            ext.gru_bwd_mfma(w["dh"], w["h_prev"], w["gates4"], self.whh_bf,
                             None, None, N, T, H,
                             w["dgi_bf"].view(N, T, 3 * H),
                             (None if self._whh_fused
                              else w["dgh_bf"].view(N, T, 3 * H)),
                             dgi_f8=w["dgi_f8"] if fp8_gru_fused else None,
                             s_dgi=self.s_dgi if fp8_gru_fused else None,
                             amax_dgi=(self.amax_dgi if fp8_gru_fused
                                       else None),
                             whh_part=(w["whh_part"] if self._whh_fused
                                       else None),
                             bhh_part=(w["bhh_part"] if self._whh_fused
                                       else None))
        elif self.bf16:
            ext.gru_bwd(w["dh"], w["h_prev"], w["gates4"], p("Whh"),
                        w["dgi"], w["dgh"], N, T, H,
                        w["dgi_bf"].view(N, T, 3 * H),
                        w["dgh_bf"].view(N, T, 3 * H))
        else:
            ext.gru_bwd(w["dh"], w["h_prev"], w["gates4"], p("Whh"),
                        w["dgi"], w["dgh"], N, T, H)
        if self.bf16:
            fork(1)
            with _on_side(self, 1):
                if self._whh_fused:
                    ext.wgrad_reduce(w["whh_part"], g("Whh"),
                                     w["bhh_part"], g("bhh"),
                                     (N + 15) // 16, False)
                else:
                    ext.gemm_tn_bf16(w["dgh_bf"].view(R, 3 * H),
                                     w["h_prev_bf"], g("Whh"), w["tn_part"],
                                     chunks, False, g("bhh"),
                                     w["tn_partb"])
                ext.gemm_tn_bf16(w["dgi_bf"].view(R, 3 * H), w["xp_bf"],
                                 g("Wih"), w["tn_part2"], chunks, False,
                                 g("bih"), w["tn_partb2"])
            if fp8_rs:
                # fp8 dgrads on the MX K=128 path (delayed per-tensor
                # scaling; the bf16 dzx copy still feeds the W1x wgrad)
                if not fp8_gru_fused:
                    ext.cast_f32_fp8_damax(w["dgi"].view(R, 3 * H),
                                           w["dgi_f8"], self.s_dgi,
                                           self.amax_dgi)
                ext.gemm_nt_fp8_rs(w["dgi_f8"], self.wihT_f8, None,
                                   self.is_wih, None, w["dzx_bf"],
                                   w["dzx_f8"], R, 3 * H, self.C, 1.0,
                                   False, inv_sa=self.is_dgi,
                                   lrelu_bwd_of=w["xp_bf"],
                                   s_out=self.s_dzx,
                                   amax_out=self.amax_dzx)
            else:
                ext.gemm_nt_bf16_rs(w["dgi_bf"].view(R, 3 * H), self.wihT_p,
                                    None, None, w["dzx_bf"], w["xp_bf"],
                                    1.0, False)
            fork(1)
            with _on_side(self, 1):
                ext.gemm_tn_bf16(w["dzx_bf"], w["xln_bf"], g("W1x"),
                                 w["tn_part3"], chunks, False, g("b1x"),
                                 w["tn_partb3"])
            if fp8_rs:
                ext.gemm_nt_fp8_rs(w["dzx_f8"], self.w1xT_f8, None,
                                   self.is_w1x, w["dxln"], None, None, R,
                                   self.C, self.C, 1.0, False,
                                   inv_sa=self.is_dzx)
            else:
                ext.gemm_nt_bf16_rs(w["dzx_bf"], self.w1xT_p, None,
                                    w["dxln"], None, None, 1.0, False)
        else:
            fork(1)
            with _on_side(self, 1):
                ext.gemm_tn(w["dgh"].view(R, 3 * H), w["h_prev"].view(R, H),
                            g("Whh"), w["tn_part"], chunks, False,
                            g("bhh"), w["tn_partb"])
                ext.gemm_tn(w["dgi"].view(R, 3 * H), w["xp"], g("Wih"),
                            w["tn_part2"], chunks, False, g("bih"),
                            w["tn_partb2"])
            ext.gemm_nn(w["dgi"].view(R, 3 * H), p("Wih"), None, w["dxp"], 1.0,
                        False, False)
            ext.lrelu_bwd(w["dxp"], w["xp"], w["dzx"])
            fork(1)
            with _on_side(self, 1):
                ext.gemm_tn(w["dzx"], w["xln"], g("W1x"), w["tn_part3"], chunks,
                            False, g("b1x"), w["tn_partb3"])
            ext.gemm_nn(w["dzx"], p("W1x"), None, w["dxln"], 1.0, False, False)
        fork()
        with _on_side(self):
            ext.ln_bwd_params(x2d, w["dxln"], w["mean"], w["rstd"],
                              w["ln_part"], g("ln_g"), g("ln_b"), chunks)
        # join: main waits for all side-stream wgrad work (+ the early
        # comm slice, so the final slice reduce and adam are ordered)
        join_streams = list(sides)
        if comm_overlap:
            join_streams.append(self.s_comm)
        for sstream in join_streams:
            if sstream is not None:
                e = torch.cuda.Event()
                e.record(sstream)
                main.wait_event(e)

    def _refresh_bf16_shadows(self):
        self.ext.cast_shadows(self.p("W1x"), self.w1x_p, self.w1xT_p,
                              self.p("Wih"), self.wih_p, self.wihT_p,
                              self.p("Whh"), self.whh_bf)
        if self.fp8:
            self.ext.absmax_scale(self.p("W1x"), self.s_w1x, self.is_w1x)
            self.ext.cast_f32_fp8_scaled(self.p("W1x"), self.w1x_f8,
                                         self.s_w1x)
            self.ext.absmax_scale(self.p("Wih"), self.s_wih, self.is_wih)
            self.ext.cast_f32_fp8_scaled(self.p("Wih"), self.wih_f8,
                                         self.s_wih)
            if self._fp8_rs:
                self.ext.cast_f32_fp8_scaled_t(self.p("Wih"), self.wihT_f8,
                                               self.s_wih)
                self.ext.cast_f32_fp8_scaled_t(self.p("W1x"), self.w1xT_f8,
                                               self.s_w1x)

    def _launch_optimizer(self, inc: bool = True):
        if inc:
            self.ext.step_inc(self.step_t)
        self.ext.adam(self.params.flat, self.grads, self.adam_m, self.adam_v,
                      self.step_t, self.lr, self.eta_min, float(self.t_max),
                      0.9, 0.999, 1e-8)
        if self.bf16:
            self._refresh_bf16_shadows()

    def _fill_rng(self, N: int):
        self.ws["eps"].normal_()
        if self.training:
            self.ws["mask"].bernoulli_(1.0 - self.DROPOUT_P)

    # max distinct (N, T) shapes whose workspaces + captured graphs stay
    # resident; beyond this the least-recently-used shape is dropped
    # (long multi-year runs see hundreds of distinct daily stock counts)
    WS_CACHE_MAX = 256

    def _ensure_ws(self, N: int, T: int):
        """Workspaces (and captured graphs) are cached per (N, T): real
        universes have a different stock count N every day, and 288 GB
        HBM3E makes keeping one ~60 MB workspace per distinct shape far
        cheaper than re-capturing the step graph each day (256 shapes
        ≈ a few tens of GB, sized for MI355X HBM).

        Overflow policy is a synchronized FULL reset — drop every
        workspace and every captured graph at once, never individual
        shapes. Destroying single hipGraphExec objects while dozens of
        sibling graphs stay live was observed to corrupt later replays
        of the SURVIVORS on ROCm 7.2 (host segfault inside
        hipGraphLaunch) once the churn got big enough: a 96-shape
        ragged-universe run crashed in epoch 2 with per-shape LRU
        eviction (profiles/r2_final_validation.md), while an
        all-at-once rebuild from an empty runtime state is the pattern
        the small-cap eviction test and every fixed-shape run already
        exercise safely. The reset costs one recapture pass (~30 ms per
        shape) and fires only when a run exceeds WS_CACHE_MAX
        (`FV_WS_CACHE` overrides) distinct day shapes."""
        if self._ws_n == N and getattr(self, "_ws_t", None) == T:
            return
        lru = self.__dict__.setdefault("_ws_lru", [])
        if (N, T) in lru:
            lru.remove((N, T))
        lru.append((N, T))
        w = self._ws_cache.get((N, T))
        if w is not None:
            self.ws = w
            self._ws_n = N
            self._ws_t = T
        else:
            cap = int(os.environ.get("FV_WS_CACHE", "0")) or self.WS_CACHE_MAX
            if len(lru) > cap:
                if self.device.type == "cuda":
                    torch.cuda.synchronize(self.device)
                self._graphs.clear()
                self._ws_cache.clear()
                lru.clear()
                lru.append((N, T))
                if not getattr(self, "_ws_reset_warned", False):
                    self._ws_reset_warned = True
                    warnings.warn(
                        f"shape-cache overflow (> {cap} distinct (N, T) "
                        "day shapes): all workspaces/graphs reset and "
                        "rebuilt; raise FV_WS_CACHE to avoid recaptures")
            self._alloc_ws(N, T)

    # -------------------------------------------------- graph capability probe
    _caps = None

    class _abort_watchdog:
        """Fail-fast guard around hipGraph captures that include RCCL
        collectives at world_size > 1: a capture hang cannot be
        interrupted in-process, so if it exceeds FV_CAPTURE_WATCHDOG_S
        (default 180 s) the rank hard-exits — torchrun then fails the
        whole job quickly instead of a multi-GPU scaling run hanging
        until the outer driver's budget is gone. No-op at ws <= 1
        (ws=1 hardware smoke: comm capture verified OK on MI355X,
        profiles/r2_rccl_ws1.md)."""

        def __init__(self, tag: str):
            self.tag = tag

        def __enter__(self):
            import threading

            self.evt = threading.Event()
            self.armed = is_distributed() and get_world_size() > 1
            if not self.armed:
                return self
            limit = float(os.environ.get("FV_CAPTURE_WATCHDOG_S", "180"))

            def _wd():
                if not self.evt.wait(limit):
                    import sys as _sys
                    print(f"[fused] {self.tag} exceeded {limit:.0f}s — "
                          f"aborting rank (set FV_COMM_GRAPH=0 to force "
                          f"the split no-comm-capture plan)",
                          file=_sys.stderr, flush=True)
                    _os._exit(17)

            self.thr = threading.Thread(target=_wd, daemon=True)
            self.thr.start()
            return self

        def __exit__(self, *a):
            if self.armed:
                self.evt.set()
            return False

    def _probe_caps(self):
        """Can torch RNG ops / RCCL collectives be captured in a hipGraph?
        Probed once; capture plans adapt (fallbacks keep correctness).

        Env guard rails (hardware de-risk for multi-GPU runs):
        - FV_COMM_GRAPH=0 forces the split plan (never attempts to
          capture an RCCL collective into a hipGraph — the safe path if
          a ROCm build hangs at comm capture);
        - FV_RNG_GRAPH=0 keeps RNG fills outside the graph.
        """
        if self._caps is not None:
            return self._caps
        allow_comm = os.environ.get("FV_COMM_GRAPH", "1") != "0"
        allow_rng = os.environ.get("FV_RNG_GRAPH", "1") != "0"
        rng_ok = allow_rng
        if rng_ok:
            try:
                t_ = torch.zeros(8, device=self.device)
                torch.cuda.synchronize()
                gp = torch.cuda.CUDAGraph()
                with torch.cuda.graph(gp):
                    t_.normal_()
                gp.replay()
                torch.cuda.synchronize()
            except Exception:
                rng_ok = False
                torch.cuda.synchronize()
        comm_ok = False
        if is_distributed() and allow_comm:
            try:
                with self._abort_watchdog("RCCL comm-capture probe"):
                    t_ = torch.ones(8, device=self.device)
                    torch.distributed.all_reduce(t_)  # eager warmup of the PG
                    torch.cuda.synchronize()
                    gp = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(gp):
                        torch.distributed.all_reduce(t_)
                    gp.replay()
                    torch.cuda.synchronize()
                comm_ok = True
            except Exception:
                comm_ok = False
                torch.cuda.synchronize()
        self._caps = (rng_ok, comm_ok)
        return self._caps

    def _graph_step_body(self, x, y, N, T, rng_in_graph: bool,
                         comm_in_graph: bool, with_opt: bool = True):
        """The full training step as a capturable kernel sequence.
        No grads.zero_(): every gradient-arena slot has exactly one
        plain-writing producer kernel per step."""
        inc_early = False
        if with_opt and self.s_side is not None:
            # LR-step counter increment runs on the side stream, overlapped
            # with forward (it must still follow the previous step's adam,
            # hence the fork event)
            e = torch.cuda.Event()
            e.record(torch.cuda.current_stream(self.device))
            self.s_side.wait_event(e)
            with torch.cuda.stream(self.s_side):
                self.ext.step_inc(self.step_t)
            inc_early = True
        if rng_in_graph:
            self._fill_rng(N)
        self._launch_forward(N, T, x=x, y=y)
        self._launch_backward(N, T, x=x, y=y, comm_overlap=comm_in_graph)
        if comm_in_graph:
            # the non-extractor slice was reduced inside backward,
            # overlapped; only the extractor slice remains
            head = self.grads[:self.params.offsets["Wenc"][0]]
            head.div_(get_world_size())
            torch.distributed.all_reduce(head)
        if with_opt:
            self._launch_optimizer(inc=not inc_early)

    def step(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        """One training step on a day cross-section; returns device loss."""
        N, T, C = x.shape
        assert C == self.C
        self._ensure_ws(N, T)
        w = self.ws
        self._main_entry()
        with self._main_ctx():
            w["x"].copy_(x)
            w["y"].copy_(y.view(N, 1))

        if not self.use_graph:
            from ..observability import roctx_range

            with self._main_ctx():
                self._fill_rng(N)
                with roctx_range("fv_forward"):
                    self._launch_forward(N, T)
                with roctx_range("fv_backward"):
                    self._launch_backward(N, T)
                if is_distributed():
                    with roctx_range("fv_allreduce"):
                        self.grads.div_(get_world_size())
                        torch.distributed.all_reduce(self.grads)
                with roctx_range("fv_adam"):
                    self._launch_optimizer()
            self._main_exit()
            return w["loss"]

        rng_ok, comm_ok = self._probe_caps()
        key = ("train", N, T)
        if key not in self._graphs:
            try:
                self._capture(key, N, T, rng_ok, comm_ok)
            except Exception:
                # capture unavailable (driver box quirk): permanent
                # fallback to the eager-launch path — same kernels, same
                # numerics, just per-launch dispatch cost
                torch.cuda.synchronize(self.device)
                self.use_graph = False
                return self.step(x, y)
        plan = self._graphs[key]
        if not rng_ok:
            with self._main_ctx():
                self._fill_rng(N)
        with self._main_ctx():
            if plan["split"]:
                plan["g_fb"].replay()
                self.grads.div_(get_world_size())
                torch.distributed.all_reduce(self.grads)
                plan["g_opt"].replay()
            else:
                plan["g"].replay()
        self._main_exit()
        return w["loss"]

    def _capture(self, key, N: int, T: int, rng_ok: bool, comm_ok: bool):
        # warmup fwd+bwd (settles lazy state; params/step counter untouched)
        torch.cuda.synchronize()
        with self._main_ctx():
            if not rng_ok:
                self._fill_rng(N)
            self.grads.zero_()
            self._launch_forward(N, T)
            self._launch_backward(N, T)
        torch.cuda.synchronize()

        split = is_distributed() and not comm_ok
        if split:
            g1 = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g1, stream=self.s_main):
                self._graph_step_body(None, None, N, T, rng_ok, False,
                                      with_opt=False)
            g2 = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g2, stream=self.s_main):
                self._launch_optimizer()
            self._graphs[key] = {"split": True, "g_fb": g1, "g_opt": g2}
        else:
            with self._abort_watchdog("train-step graph capture"):
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g, stream=self.s_main):
                    self._graph_step_body(None, None, N, T, rng_ok,
                                          is_distributed() and comm_ok)
            self._graphs[key] = {"split": False, "g": g}

    # -------------------------------------------------------- bench fast path
    def make_bench_runner(self, days):
        """Capture len(days) full training steps into ONE graph that reads
        the resident day tensors directly (no copies, no per-step host
        work). Returns (replay_fn, steps_per_replay). Falls back to the
        per-step path when collectives cannot be captured."""
        G = len(days)
        N, T, C = days[0][0].shape
        self._ensure_ws(N, T)
        rng_ok, comm_ok = self._probe_caps()
        days = [(x, y.view(-1, 1)) for x, y in days]

        if is_distributed() and not comm_ok:
            def run_fallback():
                for x, y in days:
                    self.step(x, y)
            return run_fallback, G

        # warmup
        torch.cuda.synchronize()
        with self._main_ctx():
            if not rng_ok:
                self._fill_rng(N)
            self.grads.zero_()
            self._launch_forward(N, T, x=days[0][0], y=days[0][1])
            self._launch_backward(N, T, x=days[0][0], y=days[0][1])
        torch.cuda.synchronize()

        # per-day RNG buffers when RNG can't live in the graph
        rng_bufs = None
        if not rng_ok:
            rng_bufs = [(torch.empty(N, device=self.device),
                         torch.empty(N, self.K, device=self.device))
                        for _ in range(G)]

        try:
            with self._abort_watchdog("bench multi-step graph capture"):
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g, stream=self.s_main):
                    for i, (x, y) in enumerate(days):
                        if rng_bufs is not None:
                            self.ws["eps"].copy_(rng_bufs[i][0])
                            if self.training:
                                self.ws["mask"].copy_(rng_bufs[i][1])
                        self._graph_step_body(x, y, N, T, rng_ok,
                                              is_distributed() and comm_ok)
        except Exception:
            torch.cuda.synchronize()

            def run_nograph():
                for x, y in days:
                    self.step(x, y)
            return run_nograph, G

        def run():
            if rng_bufs is not None:
                for e, m in rng_bufs:
                    e.normal_()
                    if self.training:
                        m.bernoulli_(1.0 - self.DROPOUT_P)
            self._main_entry()  # order replay after the RNG fills
            with self._main_ctx():
                g.replay()
            self._main_exit()   # next call's fills wait for this replay

        return run, G

    def forward_only(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        """Fused validation forward (no grad); returns device loss.
        hipGraph-captured per (N, T, dropout-mode) like the train step —
        the per-epoch validation pass otherwise pays full launch
        latency for every kernel."""
        N, T, C = x.shape
        self._ensure_ws(N, T)
        w = self.ws
        self._main_entry()
        with self._main_ctx():
            w["x"].copy_(x)
            w["y"].copy_(y.view(N, 1))
            self._fill_rng(N)
        if not self.use_graph:
            with self._main_ctx():
                self._launch_forward(N, T)
            self._main_exit()
            return w["loss"]
        key = ("val", N, T, self.training)
        if key not in self._graphs:
            try:
                torch.cuda.synchronize(self.device)
                with self._main_ctx():
                    self._launch_forward(N, T)  # warmup
                torch.cuda.synchronize(self.device)
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g, stream=self.s_main):
                    self._launch_forward(N, T)
                self._graphs[key] = {"g": g}
            except Exception:
                torch.cuda.synchronize(self.device)
                self._graphs[key] = {"g": None}
        g = self._graphs[key]["g"]
        with self._main_ctx():
            if g is None:
                self._launch_forward(N, T)
            else:
                g.replay()
        self._main_exit()
        return w["loss"]

    def _launch_predict(self, N: int, T: int):
        """Prediction kernel sequence: extractor -> predictor (prior) ->
        decoder with PRIOR mu/sigma (module.py:273-278)."""
        ext, p, w = self.ext, self.p, self.ws
        self._launch_forward(N, T, with_loss=False)
        ext.dec_fwd(w["h"], p("W1d"), p("b1d"), p("wmu_d"), p("bmu_d"),
                    p("wsig_d"), p("bsig_d"), p("Wb"), p("bb"), w["pmu"],
                    w["psig_c"], w["eps"], w["recon"], w["a1"], w["beta"],
                    w["asig_pre"], w["sigma"])

    def predict(self, x: torch.Tensor) -> torch.Tensor:
        """model.prediction through the fused kernels; the sequence is
        hipGraph-captured per (N, T) shape (eps refilled outside)."""
        N, T, C = x.shape
        self._ensure_ws(N, T)
        w = self.ws
        self._main_entry()
        with self._main_ctx():
            w["x"].copy_(x)
        was_training = self.training
        self.training = False  # prediction: dropout off
        try:
            with self._main_ctx():
                self._fill_rng(N)
            if not self.use_graph:
                with self._main_ctx():
                    w["y"].zero_()
                    self._launch_predict(N, T)
                self._main_exit()
                return w["recon"].view(N, 1).clone()
            key = ("predict", N, T)
            if key not in self._graphs:
                try:
                    w["y"].zero_()
                    torch.cuda.synchronize(self.device)
                    with self._main_ctx():
                        self._launch_predict(N, T)  # warmup
                    torch.cuda.synchronize(self.device)
                    gp = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(gp, stream=self.s_main):
                        self._launch_predict(N, T)
                    self._graphs[key] = {"g": gp}
                except Exception:
                    torch.cuda.synchronize(self.device)
                    self._graphs[key] = {"g": None}
            gp = self._graphs[key]["g"]
            with self._main_ctx():
                if gp is None:
                    self._launch_predict(N, T)
                else:
                    gp.replay()
            self._main_exit()
            return w["recon"].view(N, 1).clone()
        finally:
            self.training = was_training

    # ------------------------------------------------------------- epochs
    # steps per multi-step training graph (real-epoch batching)
    TRAIN_GRAPH_STEPS = 8

    def _get_multi_graph(self, N: int, T: int, g_len: int):
        """A captured graph of g_len full training steps reading from
        per-slot input buffers, with in-graph per-step loss
        accumulation into `lsum` — one replay per g_len days removes
        the per-day replay/launch host overhead from real epochs."""
        key = ("trainG", N, T, g_len)
        plan = self._graphs.get(key)
        if plan is not None:
            return plan
        w = self.ws
        xb = [torch.empty_like(w["x"]) for _ in range(g_len)]
        yb = [torch.empty_like(w["y"]) for _ in range(g_len)]
        lsum = torch.zeros(1, device=self.device)
        # warmup (params/step counter untouched)
        torch.cuda.synchronize(self.device)
        with self._main_ctx():
            self._fill_rng(N)
            self._launch_forward(N, T, x=xb[0], y=yb[0])
            self._launch_backward(N, T, x=xb[0], y=yb[0])
        torch.cuda.synchronize(self.device)
        try:
            with self._abort_watchdog("train multi-step graph capture"):
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g, stream=self.s_main):
                    for i in range(g_len):
                        self._graph_step_body(xb[i], yb[i], N, T, True,
                                              is_distributed())
                        lsum += w["loss"]
        except Exception:
            torch.cuda.synchronize(self.device)
            plan = {"g": None}
            self._graphs[key] = plan
            return plan
        plan = {"g": g, "xb": xb, "yb": yb, "lsum": lsum}
        self._graphs[key] = plan
        return plan

    def train_epoch(self, days, shuffle_order=None) -> float:
        """One epoch over device-resident day tensors. Runs of
        TRAIN_GRAPH_STEPS consecutive same-shape days execute as ONE
        multi-step graph replay (in-graph RNG + loss accumulation);
        ragged leftovers fall back to the per-day step graph."""
        total = torch.zeros((), device=self.device)
        n = 0
        days = list(days)
        G = self.TRAIN_GRAPH_STEPS
        rng_ok, comm_ok = (self._probe_caps()
                           if (self.use_graph and self.device.type == "cuda")
                           else (False, False))
        use_multi = (self.use_graph and rng_ok and G > 1
                     and not (is_distributed() and not comm_ok)
                     and len(days) > 0
                     and days[0][0].numel() * 4 <= 16 << 20)
        # (size gate: the per-slot D2D input copies must stay well under
        # the ~20 us/step host replay overhead the batching removes —
        # big A-share days fall back to per-day replays)
        i = 0
        while i < len(days):
            x, y = days[i]
            run = 1
            if use_multi:
                while (run < G and i + run < len(days)
                       and days[i + run][0].shape == x.shape):
                    run += 1
            if use_multi and run == G:
                N, T, C = x.shape
                self._ensure_ws(N, T)
                plan = self._get_multi_graph(N, T, G)
                if plan["g"] is not None:
                    self._main_entry()
                    with self._main_ctx():
                        for j in range(G):
                            plan["xb"][j].copy_(days[i + j][0])
                            plan["yb"][j].copy_(days[i + j][1].view(-1, 1))
                        plan["lsum"].zero_()
                        plan["g"].replay()
                        total += plan["lsum"][0]
                    self._main_exit()
                    n += G
                    i += G
                    continue
            loss = self.step(x, y)
            total += loss[0]
            n += 1
            i += 1
        return (total / max(n, 1)).item()

    # ------------------------------------------------- optimizer side-car
    def opt_state_dict(self) -> Dict:
        """Optimizer/scheduler state for true resume (the reference saves
        only model weights, /root/reference/main.py:79 — SURVEY.md §5.4
        calls out the side-car as a new-engine addition)."""
        return {
            "adam_m": self.adam_m.detach().cpu(),
            "adam_v": self.adam_v.detach().cpu(),
            "step_t": self.step_t.detach().cpu(),
            "lr": self.lr,
            "eta_min": self.eta_min,
            "t_max": self.t_max,
        }

    def load_opt_state_dict(self, sd: Dict) -> None:
        self.adam_m.copy_(sd["adam_m"].to(self.device))
        self.adam_v.copy_(sd["adam_v"].to(self.device))
        self.step_t.copy_(sd["step_t"].to(self.device))
        self.lr = float(sd.get("lr", self.lr))
        self.eta_min = float(sd.get("eta_min", self.eta_min))
        self.t_max = int(sd.get("t_max", self.t_max))
        # lr/t_max are baked into captured graphs as kernel args
        self._graphs.clear()
        if self.bf16:  # weights typically re-loaded alongside opt state
            self._refresh_bf16_shadows()

    @torch.no_grad()
    def validate_epoch(self, days) -> float:
        was = self.training
        self.training = False
        total = torch.zeros((), device=self.device)
        n = 0
        try:
            for x, y in days:
                loss = self.forward_only(x, y)
                total += loss[0]
                n += 1
        finally:
            self.training = was
        return (total / max(n, 1)).item()
