"""Done-masked multi-layer LSTM core.

Reference semantics: AtariNet's per-step unroll with done-masking
(algorithms/utils/atari_model.py:109-120) — `state *= notdone` each step,
then one LSTM step.  The reference calls nn.LSTM per step from Python; here:

- input-side GEMM for ALL T steps is hoisted into one rocBLAS GEMM
  ([T*B, in] @ W_ih^T);
- the sequential loop does one rocBLAS GEMM (h @ W_hh^T) + ONE fused HIP
  pointwise kernel per step (csrc/lstm.hip), forward and backward;
- CPU path uses the same structure with torch ops (the oracle).

Weights are plain nn.LSTM-layout parameters (weight_ih_l{k} etc.) so
checkpoints interop with torch LSTMs.
"""

from __future__ import annotations

import ctypes
from typing import List, Tuple

import torch
import torch.nn as nn

from . import _backend

_c = ctypes.c_void_p


def _pointwise_fwd(gates: torch.Tensor, c_prev: torch.Tensor,
                   out_h: torch.Tensor = None, out_c: torch.Tensor = None):
    """Returns (h, c); `gates` is overwritten with activated gates.  When
    ``out_h/out_c`` are given the results are written there (one fewer
    launch per step on the hot path)."""
    B, H4 = gates.shape
    H = H4 // 4
    if gates.is_cuda:
        h = out_h if out_h is not None else torch.empty(
            B, H, device=gates.device, dtype=torch.float32)
        c = out_c if out_c is not None else torch.empty_like(h)
        ret = _backend.lib().lstm_pointwise_fwd(
            _c(gates.data_ptr()), _c(c_prev.data_ptr()), _c(h.data_ptr()),
            _c(c.data_ptr()), B, H, _backend.current_stream())
        _backend.check(ret, "lstm_pointwise_fwd")
        return h, c
    i, f, g, o = gates.chunk(4, dim=1)
    i, f, g, o = i.sigmoid(), f.sigmoid(), g.tanh(), o.sigmoid()
    c_new = f * c_prev + i * g
    h_new = o * torch.tanh(c_new)
    gates.copy_(torch.cat([i, f, g, o], dim=1))  # match GPU in-place contract
    if out_h is not None:
        out_h.copy_(h_new)
        h_new = out_h
    if out_c is not None:
        out_c.copy_(c_new)
        c_new = out_c
    return h_new, c_new


def _pointwise_bwd(gates_act, c_prev, c_out, dh, dc_in,
                   out_dgates: torch.Tensor = None,
                   out_dc_prev: torch.Tensor = None):
    B, H4 = gates_act.shape
    H = H4 // 4
    if gates_act.is_cuda:
        dgates = out_dgates if out_dgates is not None else \
            torch.empty_like(gates_act)
        dc_prev = out_dc_prev if out_dc_prev is not None else torch.empty(
            B, H, device=gates_act.device, dtype=torch.float32)
        ret = _backend.lib().lstm_pointwise_bwd(
            _c(gates_act.data_ptr()), _c(c_prev.data_ptr()), _c(c_out.data_ptr()),
            _c(dh.data_ptr()), _c(dc_in.data_ptr()) if dc_in is not None else None,
            _c(dgates.data_ptr()), _c(dc_prev.data_ptr()), B, H,
            _backend.current_stream())
        _backend.check(ret, "lstm_pointwise_bwd")
        return dgates, dc_prev
    i, f, g, o = gates_act.chunk(4, dim=1)
    tc = torch.tanh(c_out)
    dc = dh * o * (1 - tc * tc)
    if dc_in is not None:
        dc = dc + dc_in
    di = dc * g * i * (1 - i)
    df = dc * c_prev * f * (1 - f)
    dg = dc * i * (1 - g * g)
    do = dh * tc * o * (1 - o)
    dgates = torch.cat([di, df, dg, do], dim=1)
    dc_prev = dc * f
    if out_dgates is not None:
        out_dgates.copy_(dgates)
        dgates = out_dgates
    if out_dc_prev is not None:
        out_dc_prev.copy_(dc_prev)
        dc_prev = out_dc_prev
    return dgates, dc_prev


class _MaskedLSTMFn(torch.autograd.Function):
    """One LSTM layer unrolled over T with per-step done masking.

    x: [T,B,I]; notdone: [T,B,1]; h0,c0: [B,H]; weights nn.LSTM layout.
    Strategy: xW^T for all T hoisted to one GEMM; per step one GEMM + one
    pointwise kernel.  Saves activated gates + cell states for backward.
    """

    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.float32)
    def forward(ctx, x, notdone, h0, c0, w_ih, w_hh, b_ih, b_hh):
        T, B, I = x.shape
        H = w_hh.shape[1]
        xg = torch.addmm(b_ih + b_hh, x.reshape(T * B, I), w_ih.t()).view(T, B, 4 * H)
        h, c = h0.contiguous(), c0.contiguous()
        hs = torch.empty(T, B, H, device=x.device, dtype=x.dtype)
        cs_in = torch.empty(T, B, H, device=x.device, dtype=x.dtype)
        cs_out = torch.empty(T, B, H, device=x.device, dtype=x.dtype)
        hs_in = torch.empty(T, B, H, device=x.device, dtype=x.dtype)
        gates_all = torch.empty(T, B, 4 * H, device=x.device, dtype=x.dtype)
        w_hh_t = w_hh.t()
        for t in range(T):
            nd = notdone[t]
            # out=-form writes land directly in the saved [T,...] buffers:
            # 4 enqueues per step instead of 8 (the unroll is launch-bound)
            h = torch.mul(h, nd, out=hs_in[t])
            c = torch.mul(c, nd, out=cs_in[t])
            torch.addmm(xg[t], h, w_hh_t, out=gates_all[t])
            h, c = _pointwise_fwd(gates_all[t], c, out_h=hs[t],
                                  out_c=cs_out[t])
        ctx.save_for_backward(x, notdone, hs_in, cs_in, cs_out, gates_all,
                              w_ih, w_hh)
        ctx.H = H
        # final h/c alias hs[T-1]/cs_out[T-1] after the out=-form loop;
        # clone so the Function's outputs don't share storage
        return hs, h.clone(), c.clone()

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, d_hs, d_hT, d_cT):
        (x, notdone, hs_in, cs_in, cs_out, gates_all, w_ih,
         w_hh) = ctx.saved_tensors
        T, B, I = x.shape
        H = ctx.H
        dh_carry = d_hT.contiguous().clone()
        dc_carry = d_cT.contiguous().clone()
        dgates_all = torch.empty_like(gates_all)
        dh_buf = torch.empty_like(dh_carry)
        dc_prev_buf = torch.empty_like(dc_carry)
        for t in range(T - 1, -1, -1):
            dh = torch.add(d_hs[t], dh_carry, out=dh_buf)
            dgates, dc_prev = _pointwise_bwd(
                gates_all[t], cs_in[t], cs_out[t], dh, dc_carry,
                out_dgates=dgates_all[t], out_dc_prev=dc_prev_buf)
            nd = notdone[t]
            torch.mm(dgates, w_hh, out=dh_carry)
            dh_carry.mul_(nd)
            torch.mul(dc_prev, nd, out=dc_carry)
        dg2 = dgates_all.reshape(T * B, 4 * H)
        dx = (dg2 @ w_ih).view(T, B, I)
        dw_ih = dg2.t() @ x.reshape(T * B, I)
        dw_hh = dg2.t() @ hs_in.reshape(T * B, H)
        db = dg2.sum(0)
        return dx, None, dh_carry, dc_carry, dw_ih, dw_hh, db, db


class _MaskedLSTMSeqFn(torch.autograd.Function):
    """One LSTM layer unrolled over T with the loop driven from C++
    (csrc/lstm_seq.hip): per step one rocBLAS SGEMM + one fused kernel, all
    enqueued natively — removes the ~20 ms/iter of Python launch overhead
    the per-step path pays (profiles/README.md).  Numerics identical to
    :class:`_MaskedLSTMFn`.  EXPERIMENTAL: enabled via SCALERL_LSTM_SEQ=1
    until hardware-validated (its oracle test is test_lstm_seq*)."""

    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.float32)
    def forward(ctx, x, notdone, h0, c0, w_ih, w_hh, b_ih, b_hh):
        lib = _backend.lib()
        T, B, I = x.shape
        H = w_hh.shape[1]
        dev = x.device
        xg = torch.addmm(b_ih + b_hh, x.reshape(T * B, I),
                         w_ih.t()).view(T, B, 4 * H).contiguous()
        h = h0.contiguous().clone()
        c = c0.contiguous().clone()
        hs = torch.empty(T, B, H, device=dev)
        hs_in = torch.empty(T, B, H, device=dev)
        cs_in = torch.empty(T, B, H, device=dev)
        cs_out = torch.empty(T, B, H, device=dev)
        gemm_tmp = torch.empty(B, 4 * H, device=dev)
        nd = notdone.reshape(T, B, 1)[:, :, 0].contiguous()
        w_hh_c = w_hh.contiguous()
        ret = lib.masked_lstm_seq_fwd(
            _cp(xg), _cp(w_hh_c), _cp(nd), _cp(h), _cp(c), _cp(hs),
            _cp(hs_in), _cp(cs_in), _cp(cs_out), _cp(gemm_tmp), T, B, H,
            _backend.current_stream())
        _backend.check(ret, "masked_lstm_seq_fwd")
        ctx.save_for_backward(x, nd, hs_in, cs_in, cs_out, xg, w_ih, w_hh_c)
        ctx.H = H
        return hs, h, c

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, d_hs, d_hT, d_cT):
        lib = _backend.lib()
        x, nd, hs_in, cs_in, cs_out, gates_all, w_ih, w_hh = ctx.saved_tensors
        T, B, I = x.shape
        H = ctx.H
        dev = x.device
        dgates_all = torch.empty(T, B, 4 * H, device=dev)
        dh_carry = d_hT.contiguous().clone()
        dc_carry = d_cT.contiguous().clone()
        dc_prev_tmp = torch.empty(B, H, device=dev)
        ret = lib.masked_lstm_seq_bwd(
            _cp(gates_all), _cp(cs_in), _cp(cs_out),
            _cp(d_hs.contiguous()), _cp(nd), _cp(w_hh), _cp(dgates_all),
            _cp(dh_carry), _cp(dc_carry), _cp(dc_prev_tmp), T, B, H,
            _backend.current_stream())
        _backend.check(ret, "masked_lstm_seq_bwd")
        dg2 = dgates_all.reshape(T * B, 4 * H)
        dx = (dg2 @ w_ih).view(T, B, I)
        dw_ih = dg2.t() @ x.reshape(T * B, I)
        dw_hh = dg2.t() @ hs_in.reshape(T * B, H)
        db = dg2.sum(0)
        return dx, None, dh_carry, dc_carry, dw_ih, dw_hh, db, db


def _cp(t: torch.Tensor):
    return _c(t.data_ptr())


_c = ctypes.c_void_p
_USE_SEQ = None


def _use_seq_path() -> bool:
    """C++-driven sequence loop (hardware-validated r2: oracle tests pass,
    25.5 vs 28.2 ms/iter at B=256 on the learner micro).  Default ON on
    GPU; SCALERL_LSTM_SEQ=0 falls back to the per-step Python path."""
    global _USE_SEQ
    if _USE_SEQ is None:
        import os
        _USE_SEQ = os.environ.get("SCALERL_LSTM_SEQ", "1") != "0"
    return _USE_SEQ


class MaskedLSTM(nn.Module):
    """Multi-layer done-masked LSTM with nn.LSTM-compatible parameters."""

    def __init__(self, input_size: int, hidden_size: int, num_layers: int = 1):
        super().__init__()
        self.input_size, self.hidden_size = input_size, hidden_size
        self.num_layers = num_layers
        for k in range(num_layers):
            in_sz = input_size if k == 0 else hidden_size
            w_ih = nn.Parameter(torch.empty(4 * hidden_size, in_sz))
            w_hh = nn.Parameter(torch.empty(4 * hidden_size, hidden_size))
            b_ih = nn.Parameter(torch.empty(4 * hidden_size))
            b_hh = nn.Parameter(torch.empty(4 * hidden_size))
            # match nn.LSTM defaults: ALL params uniform(-H^-0.5, H^-0.5),
            # biases included, so fresh-training dynamics match the
            # reference's torch LSTM core (atari_model.py:51-55)
            for w in (w_ih, w_hh, b_ih, b_hh):
                nn.init.uniform_(w, -hidden_size ** -0.5, hidden_size ** -0.5)
            setattr(self, f"weight_ih_l{k}", w_ih)
            setattr(self, f"weight_hh_l{k}", w_hh)
            setattr(self, f"bias_ih_l{k}", b_ih)
            setattr(self, f"bias_hh_l{k}", b_hh)

    def initial_state(self, batch_size: int, device=None):
        z = torch.zeros(self.num_layers, batch_size, self.hidden_size,
                        device=device)
        return (z, z.clone())

    def forward(self, x: torch.Tensor, notdone: torch.Tensor,
                state: Tuple[torch.Tensor, torch.Tensor]):
        """x [T,B,I], notdone [T,B] → out [T,B,H], (hN, cN) [L,B,H]."""
        h0, c0 = state
        # The recurrent core runs fp32 (it is tiny and latency-bound; the
        # fused pointwise kernels are fp32) even when the encoder is bf16.
        x = x.float()
        nd = notdone.unsqueeze(-1).float()
        hs_out: List[torch.Tensor] = []
        cs_out: List[torch.Tensor] = []
        out = x
        # The C++ seq loop drives a process-global rocBLAS handle, which is
        # NOT safe to call from two threads at once (the thread-mode
        # inference worker shares the process with the learner).  T==1 has
        # no launch-overhead to amortize anyway, so single-step forwards
        # (the inference worker's shape) take the per-step path, keeping
        # the seq loop learner-only.
        fn = (_MaskedLSTMSeqFn if (x.is_cuda and x.shape[0] > 1 and
                                   _use_seq_path())
              else _MaskedLSTMFn)
        for k in range(self.num_layers):
            out, hN, cN = fn.apply(
                out.contiguous(), nd, h0[k], c0[k],
                getattr(self, f"weight_ih_l{k}"),
                getattr(self, f"weight_hh_l{k}"),
                getattr(self, f"bias_ih_l{k}"),
                getattr(self, f"bias_hh_l{k}"))
            hs_out.append(hN)
            cs_out.append(cN)
        return out, (torch.stack(hs_out), torch.stack(cs_out))
