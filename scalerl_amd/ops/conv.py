"""Hand-written MFMA Atari conv encoder (fwd + wgrad + dgrad wrappers).

All v2 kernels are hardware-validated (r2: 10/10 kernel-vs-oracle on
MI355X, tests ungated).  MIOpen remains the model default until the v3
kernels (panel-staged fwd/wgrad, stride-decomposed dgrad — written,
CPU-math-verified, gated SCALERL_EXPERIMENTAL) win the per-op A/B
(profiles/README.md).  `mfma_selftest` validates the fragment-layout
constants first; every kernel shares them.
"""

from __future__ import annotations

import ctypes

import torch

from . import _backend

_c = ctypes.c_void_p


def _declare_conv(lib):
    if getattr(lib, "_conv_declared", False):
        return lib
    c = ctypes
    for name in ("atari_conv1_fwd_u8", "atari_conv1_fwd_bf16",
                 "atari_conv2_fwd", "atari_conv3_fwd",
                 "atari_conv2_fwd_v3", "atari_conv3_fwd_v3"):
        fn = getattr(lib, name)
        fn.argtypes = [c.c_void_p, c.c_void_p, c.c_void_p, c.c_void_p,
                       c.c_long, c.c_int, c.c_void_p]
        fn.restype = c.c_int
    lib.mfma_selftest.argtypes = [c.c_void_p, c.c_void_p, c.c_void_p,
                                  c.c_void_p]
    lib.mfma_selftest.restype = c.c_int
    lib._conv_declared = True
    return lib


def mfma_selftest() -> bool:
    """Validate the 16x16x32 bf16 fragment maps on-device."""
    lib = _declare_conv(_backend.lib())
    g = torch.Generator().manual_seed(0)
    A = torch.randn(16, 32, generator=g).cuda()
    B = torch.randn(32, 16, generator=g).cuda()  # asymmetric (guide G9)
    D = torch.empty(16, 16, device="cuda")
    ret = lib.mfma_selftest(_c(A.data_ptr()), _c(B.data_ptr()),
                            _c(D.data_ptr()), _backend.current_stream())
    _backend.check(ret, "mfma_selftest")
    want = (A.to(torch.bfloat16).float() @ B.to(torch.bfloat16).float())
    return torch.allclose(D.cpu(), want.cpu(), rtol=1e-2, atol=1e-2)


_SHAPES = {
    1: ("atari_conv1_fwd", (4, 84, 84), (32, 20, 20)),
    2: ("atari_conv2_fwd", (32, 20, 20), (64, 9, 9)),
    3: ("atari_conv3_fwd", (64, 9, 9), (64, 7, 7)),
}


@torch.no_grad()
def atari_conv_fwd(layer: int, x: torch.Tensor, weight: torch.Tensor,
                   bias: torch.Tensor = None, relu: bool = True) -> torch.Tensor:
    """Forward one encoder conv.  layer 1 accepts uint8 (fused /255) or
    bf16; layers 2-3 take bf16.  Returns bf16 [N, K, OH, OW]."""
    name, in_shape, out_shape = _SHAPES[layer]
    assert tuple(x.shape[1:]) == in_shape, (x.shape, in_shape)
    lib = _declare_conv(_backend.lib())
    if layer == 1:
        fn = (lib.atari_conv1_fwd_u8 if x.dtype == torch.uint8
              else lib.atari_conv1_fwd_bf16)
        if x.dtype != torch.uint8:
            x = x.to(torch.bfloat16)
    else:
        fn = getattr(lib, name)
        x = x.to(torch.bfloat16)
    w = weight.to(torch.bfloat16).contiguous()
    b = bias.float().contiguous() if bias is not None else None
    out = torch.empty((x.shape[0], *out_shape), dtype=torch.bfloat16,
                      device=x.device)
    xc = x.contiguous()  # local outlives the launch (see atari_conv_wgrad)
    ret = fn(_c(xc.data_ptr()), _c(w.data_ptr()),
             _c(b.data_ptr()) if b is not None else None,
             _c(out.data_ptr()), x.shape[0], int(relu),
             _backend.current_stream())
    _backend.check(ret, name)
    del xc
    return out


def _declare_bwd(lib):
    if getattr(lib, "_conv_bwd_declared", False):
        return lib
    c = ctypes
    for name in ("atari_conv1_wgrad_u8", "atari_conv2_wgrad",
                 "atari_conv3_wgrad"):
        fn = getattr(lib, name)
        fn.argtypes = [c.c_void_p, c.c_void_p, c.c_void_p, c.c_long,
                       c.c_long, c.c_void_p]
        fn.restype = c.c_int
    for name in ("atari_conv2_dgrad", "atari_conv3_dgrad",
                 "atari_conv2_dgrad_v3", "atari_conv1_wgrad_v3",
                 "atari_conv2_wgrad_v3", "atari_conv3_wgrad_v3"):
        fn = getattr(lib, name)
        fn.argtypes = [c.c_void_p, c.c_void_p, c.c_void_p, c.c_long,
                       c.c_void_p]
        fn.restype = c.c_int
    lib._conv_bwd_declared = True
    return lib


_WGRAD = {1: "atari_conv1_wgrad_u8", 2: "atari_conv2_wgrad",
          3: "atari_conv3_wgrad"}
_DGRAD = {2: "atari_conv2_dgrad", 3: "atari_conv3_dgrad"}
_WSHAPE = {1: (32, 4, 8, 8), 2: (64, 32, 4, 4), 3: (64, 64, 3, 3)}


@torch.no_grad()
def atari_conv_wgrad(layer: int, x: torch.Tensor, dout: torch.Tensor,
                     split: int = 64) -> torch.Tensor:
    """dL/dW for one encoder conv (fp32 out).  Layer 1 takes uint8 x.

    NOTE every cast/contiguous result is BOUND TO A LOCAL that outlives
    the launch: taking ``.data_ptr()`` of an unreferenced temporary frees
    its block before ``fn`` is even called, and the NEXT argument's
    allocation can reuse and overwrite it (this exact use-after-free
    corrupted dgrad results for weeks — r2 debug log, profiles/README.md)."""
    lib = _declare_bwd(_declare_conv(_backend.lib()))
    K, C, KH, KW = _WSHAPE[layer]
    dw = torch.zeros(K, C * KH * KW, device=x.device, dtype=torch.float32)
    if layer == 1:
        assert x.dtype == torch.uint8
        xc = x.contiguous()
    else:
        xc = x.to(torch.bfloat16).contiguous()
    dc = dout.to(torch.bfloat16).contiguous()
    fn = getattr(lib, _WGRAD[layer])
    ret = fn(_c(xc.data_ptr()), _c(dc.data_ptr()),
             _c(dw.data_ptr()), x.shape[0], split, _backend.current_stream())
    _backend.check(ret, _WGRAD[layer])
    del xc, dc  # keep alive past the (async, same-stream) launch
    return dw.view(K, C, KH, KW)


@torch.no_grad()
def atari_conv_dgrad(layer: int, dout: torch.Tensor,
                     weight: torch.Tensor) -> torch.Tensor:
    """dL/dX for encoder convs 2-3 (bf16 out; conv1 is the input layer).
    Temp-lifetime discipline: see atari_conv_wgrad."""
    lib = _declare_bwd(_declare_conv(_backend.lib()))
    _, in_shape, _ = _SHAPES[layer]
    din = torch.empty((dout.shape[0], *in_shape), dtype=torch.bfloat16,
                      device=dout.device)
    dc = dout.to(torch.bfloat16).contiguous()
    wc = weight.to(torch.bfloat16).contiguous()
    fn = getattr(lib, _DGRAD[layer])
    ret = fn(_c(dc.data_ptr()), _c(wc.data_ptr()),
             _c(din.data_ptr()), dout.shape[0], _backend.current_stream())
    _backend.check(ret, _DGRAD[layer])
    del dc, wc
    return din


@torch.no_grad()
def atari_conv2_dgrad_v3(dout: torch.Tensor,
                         weight: torch.Tensor) -> torch.Tensor:
    """Parity-decomposed stride-2 dgrad for conv2: 4x less MFMA work than
    the masked v2 (csrc/conv_bwd.hip conv2_dgrad_v3).  EXPERIMENTAL until
    hardware-validated (r3); temp-lifetime discipline as in
    atari_conv_wgrad."""
    lib = _declare_bwd(_declare_conv(_backend.lib()))
    din = torch.empty((dout.shape[0], 32, 20, 20), dtype=torch.bfloat16,
                      device=dout.device)
    dc = dout.to(torch.bfloat16).contiguous()
    wc = weight.to(torch.bfloat16).contiguous()
    ret = lib.atari_conv2_dgrad_v3(
        _c(dc.data_ptr()), _c(wc.data_ptr()), _c(din.data_ptr()),
        dout.shape[0], _backend.current_stream())
    _backend.check(ret, "atari_conv2_dgrad_v3")
    del dc, wc
    return din


_WGRAD_V3 = {1: "atari_conv1_wgrad_v3", 2: "atari_conv2_wgrad_v3",
             3: "atari_conv3_wgrad_v3"}


@torch.no_grad()
def atari_conv_fwd_v3(layer: int, x: torch.Tensor, weight: torch.Tensor,
                      bias: torch.Tensor = None,
                      relu: bool = True) -> torch.Tensor:
    """Panel-staged forward for conv2/conv3 (conv_fwd.hip convN_fwd_v3).
    EXPERIMENTAL until hardware-validated (r3)."""
    assert layer in (2, 3)
    name, in_shape, out_shape = _SHAPES[layer]
    assert tuple(x.shape[1:]) == in_shape
    lib = _declare_conv(_backend.lib())
    fn = getattr(lib, f"atari_conv{layer}_fwd_v3")
    xc = x.to(torch.bfloat16).contiguous()
    w = weight.to(torch.bfloat16).contiguous()
    b = bias.float().contiguous() if bias is not None else None
    out = torch.empty((x.shape[0], *out_shape), dtype=torch.bfloat16,
                      device=x.device)
    ret = fn(_c(xc.data_ptr()), _c(w.data_ptr()),
             _c(b.data_ptr()) if b is not None else None,
             _c(out.data_ptr()), x.shape[0], int(relu),
             _backend.current_stream())
    _backend.check(ret, f"atari_conv{layer}_fwd_v3")
    del xc, w, b
    return out


@torch.no_grad()
def atari_conv_wgrad_v3(layer: int, x: torch.Tensor,
                        dout: torch.Tensor) -> torch.Tensor:
    """Panel-staged wgrad (conv_bwd.hip convN_wgrad_v3): both MFMA
    operands are contiguous LDS vector reads (v2's B side re-gathered
    im2col scalar-by-scalar per tile).  EXPERIMENTAL until
    hardware-validated (r3)."""
    lib = _declare_bwd(_declare_conv(_backend.lib()))
    K, C, KH, KW = _WSHAPE[layer]
    dw = torch.zeros(K, C * KH * KW, device=x.device, dtype=torch.float32)
    if layer == 1:
        assert x.dtype == torch.uint8
        xc = x.contiguous()
    else:
        xc = x.to(torch.bfloat16).contiguous()
    dc = dout.to(torch.bfloat16).contiguous()
    fn = getattr(lib, _WGRAD_V3[layer])
    ret = fn(_c(xc.data_ptr()), _c(dc.data_ptr()), _c(dw.data_ptr()),
             x.shape[0], _backend.current_stream())
    _backend.check(ret, _WGRAD_V3[layer])
    del xc, dc
    return dw.view(K, C, KH, KW)


class _NativeConvFn(torch.autograd.Function):
    """Autograd over the native encoder convs (layer 2/3: full backward;
    layer 1: wgrad only — it is the input layer)."""

    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda")
    def forward(ctx, x, weight, bias, layer, relu):
        out = atari_conv_fwd(layer, x, weight, bias, relu=relu)
        ctx.save_for_backward(x, weight, out)
        ctx.layer, ctx.relu = layer, relu
        return out

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, dout):
        x, weight, out = ctx.saved_tensors
        layer = ctx.layer
        if ctx.relu:
            dout = dout * (out > 0)
        dw = atari_conv_wgrad(layer, x, dout)
        db = dout.float().sum(dim=(0, 2, 3))
        dx = None
        if layer in _DGRAD and ctx.needs_input_grad[0]:
            dx = atari_conv_dgrad(layer, dout, weight)
        return dx, dw, db, None, None


def native_conv(layer: int, x, weight, bias=None, relu: bool = True):
    """Differentiable native encoder conv (EXPERIMENTAL, opt-in via
    AtariNet(native_conv=True) / SCALERL_NATIVE_CONV=1)."""
    return _NativeConvFn.apply(x, weight, bias, layer, relu)
