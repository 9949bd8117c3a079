"""lpx_torch — fp8 (OCP e4m3) ResNet-50 inference on gfx950 MFMA
(recipe: torch; BASELINE.json config 5).

Every conv/fc runs on the hand-written fp8 MFMA GEMM kernel (gemm_fp8 in
lpx_ops): 1x1 convs are direct GEMMs over NHWC rows; 3x3/7x7 go through the
fused im2col+quantize kernel. BN is folded into conv scale/bias at init
(inference). Weights are quantized to e4m3 once at load (per-tensor scale =
amax/448); activations are quantized per call with a dynamic per-tensor
scale. Residual adds / pooling are cheap elementwise torch ops — the MFMA
GEMMs are the hot path rocprof must show.
"""

from __future__ import annotations

import math
from typing import List, Optional, Tuple

import torch
import torch.nn.functional as F

try:
    import lpx_ops as _ops
except ImportError:
    from lambdipy_amd import ops as _ops

E4M3_MAX = 448.0


def _quant_weight(w: torch.Tensor) -> Tuple[torch.Tensor, float]:
    """w [KO, K] f32 -> (e4m3 bytes, scale)."""
    amax = float(w.abs().amax().clamp(min=1e-8))
    scale = amax / E4M3_MAX
    return _ops.ext().quantize_fp8(w.contiguous(), scale), scale


def _act_scale(x: torch.Tensor) -> float:
    return float(x.abs().amax().clamp(min=1e-8)) / E4M3_MAX


# Activation scales are calibrated on the FIRST batch and then frozen: the
# dynamic amax is a device->host sync per conv (53 per forward) and dominates
# small-batch latency. Frozen scales get 2x headroom against later batches.
_CAL_HEADROOM = 2.0


class Conv2dFP8:
    """NHWC conv as fp8 MFMA GEMM (+ folded-BN bias, optional fused relu)."""

    def __init__(self, cin: int, cout: int, ksize: int, stride: int, pad: int,
                 relu: bool, device, gen: torch.Generator):
        self.cin, self.cout = cin, cout
        self.k, self.stride, self.pad = ksize, stride, pad
        self.relu = relu
        fan_in = cin * ksize * ksize
        # generate on the GENERATOR's device (a cuda generator makes the
        # 25M-param ResNet init ~1 s faster than CPU randn + H2D)
        gdev = gen.device if gen is not None else torch.device(device)
        w = torch.randn(cout, ksize * ksize * cin, generator=gen,
                        dtype=torch.float32, device=gdev)
        if gdev != torch.device(device):
            w = w.to(device)
        w *= 1.0 / math.sqrt(fan_in)
        self.wq, self.wscale = _quant_weight(w)
        b = torch.randn(cout, generator=gen, dtype=torch.float32, device=gdev)
        self.bias = (b.to(device) if gdev != torch.device(device) else b) * 0.01
        self.sa: Optional[float] = None  # frozen activation scale

    def out_shape(self, H: int, W: int) -> Tuple[int, int]:
        ho = (H + 2 * self.pad - self.k) // self.stride + 1
        wo = (W + 2 * self.pad - self.k) // self.stride + 1
        return ho, wo

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        # x: [N, H, W, Cin] f32
        N, H, W, C = x.shape
        e = _ops.ext()
        if self.sa is None:
            self.sa = _act_scale(x) * _CAL_HEADROOM
        sa = self.sa
        ho, wo = self.out_shape(H, W)
        if self.k == 1 and self.stride == 1 and self.pad == 0:
            aq = e.quantize_fp8(x.reshape(-1, C).contiguous(), sa)
            out = e.gemm_fp8(aq, self.wq, sa * self.wscale, self.bias, self.relu)
        elif C % 16 == 0:
            # implicit GEMM: quantize NHWC once, gather patches in staging
            xq = e.quantize_fp8(x.contiguous(), sa)
            out = e.conv_fp8(xq, self.wq, H, W, C, self.k, self.k, self.stride,
                             self.pad, sa * self.wscale, self.bias, self.relu)
        else:
            # first layer (C=3): materialized im2col patch matrix
            aq = e.im2col_fp8(x.contiguous(), self.k, self.k, self.stride,
                              self.pad, sa)
            out = e.gemm_fp8(aq, self.wq, sa * self.wscale, self.bias, self.relu)
        return out.view(N, ho, wo, self.cout)

    # -- fused-chain entry points (inputs/outputs as e4m3 bytes) -----------
    # xq is NHWC e4m3 quantized at self.sa (the producer wrote it with
    # out_scale = self.sa). out_scale=None -> f32 result; else e4m3 bytes.
    def run_q(self, xq: torch.Tensor, N: int, H: int, W: int,
              out_scale: Optional[float],
              resid: Optional[torch.Tensor] = None,
              relu: Optional[bool] = None):
        e = _ops.ext()
        desc = self.sa * self.wscale
        ho, wo = self.out_shape(H, W)
        if relu is None:
            relu = self.relu
        if self.k == 1 and self.stride == 1 and self.pad == 0:
            a2 = xq.reshape(-1, self.cin)
            if out_scale is None:
                out = e.gemm_fp8(a2, self.wq, desc, self.bias, relu,
                                 resid.reshape(-1, self.cout).contiguous()
                                 if resid is not None else None)
            else:
                out = e.gemm_fp8_q8(a2, self.wq, desc, self.bias, relu,
                                    out_scale)
        else:
            if out_scale is None:
                out = e.conv_fp8(xq, self.wq, H, W, self.cin, self.k, self.k,
                                 self.stride, self.pad, desc, self.bias,
                                 relu)
            else:
                out = e.conv_fp8_q8(xq, self.wq, H, W, self.cin, self.k,
                                    self.k, self.stride, self.pad, desc,
                                    self.bias, relu, out_scale)
        return out.view(N, ho, wo, self.cout), ho, wo


class Bottleneck:
    def __init__(self, cin: int, mid: int, cout: int, stride: int, device, gen):
        self.conv1 = Conv2dFP8(cin, mid, 1, 1, 0, True, device, gen)
        self.conv2 = Conv2dFP8(mid, mid, 3, stride, 1, True, device, gen)
        self.conv3 = Conv2dFP8(mid, cout, 1, 1, 0, False, device, gen)
        self.down: Optional[Conv2dFP8] = None
        if stride != 1 or cin != cout:
            self.down = Conv2dFP8(cin, cout, 1, stride, 0, False, device, gen)

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        if self.conv2.sa is not None and self.conv3.sa is not None:
            # fused chain: conv1 and conv2 emit e4m3 at the NEXT conv's
            # frozen input scale — skips two f32 write/read/quantize passes
            e = _ops.ext()
            N, H, W, C = x.shape
            xq = e.quantize_fp8(x.contiguous(), self.conv1.sa)
            if self.down is None:
                idn = x
            else:
                # conv1 and down calibrate on the SAME tensor, so their
                # frozen scales are identical — reuse the quantized bytes
                idn, _, _ = self.down.run_q(xq, N, H, W, None)
            q1, h, w = self.conv1.run_q(xq, N, H, W, self.conv2.sa)
            q2, h, w = self.conv2.run_q(q1.contiguous(), N, h, w, self.conv3.sa)
            # conv3 fuses the identity add + relu into its epilogue when the
            # shape hits the templated scaled-MFMA path (RES instantiation);
            # otherwise keep the elementwise add (a runtime epilogue branch
            # in the small-shape kernels measured slower than the EW pass)
            c3 = self.conv3
            fuse_res = (N * h * w) % 256 == 0 and c3.cout % 256 == 0 and \
                c3.cin % 128 == 0
            if fuse_res:
                y, h, w = c3.run_q(q2.contiguous(), N, h, w, None,
                                   resid=idn, relu=True)
                return y
            y, h, w = c3.run_q(q2.contiguous(), N, h, w, None)
            return F.relu(y + idn)
        idn = x if self.down is None else self.down(x)
        y = self.conv3(self.conv2(self.conv1(x)))  # calibration pass
        return F.relu(y + idn)


class ResNet50FP8:
    """ResNet-50 v1 inference graph, random-init (no network for weights)."""

    LAYERS = [(64, 256, 3, 1), (128, 512, 4, 2), (256, 1024, 6, 2),
              (512, 2048, 3, 2)]

    def __init__(self, n_classes: int = 1000, device: str = "cuda", seed: int = 0):
        self.device = torch.device(device)
        # device generator: randn for the 25M parameters runs on the GPU
        # (CPU gen costs ~1 s of cold start at model_build)
        try:
            gen = torch.Generator(device=self.device).manual_seed(seed)
        except RuntimeError:
            gen = torch.Generator().manual_seed(seed)
        self.conv1 = Conv2dFP8(3, 64, 7, 2, 3, True, self.device, gen)
        self.blocks: List[Bottleneck] = []
        cin = 64
        for mid, cout, n, stride in self.LAYERS:
            for i in range(n):
                self.blocks.append(
                    Bottleneck(cin, mid, cout, stride if i == 0 else 1,
                               self.device, gen))
                cin = cout
        fcw = torch.randn(n_classes, 2048, generator=gen,
                          device=gen.device) * (1.0 / math.sqrt(2048))
        self.fcq, self.fcscale = _quant_weight(fcw.to(self.device))
        self.fc_bias = torch.zeros(n_classes, dtype=torch.float32,
                                   device=self.device)
        self.fc_sa: Optional[float] = None
        self.n_classes = n_classes
        self._graph = None
        self._static_in = None
        self._static_out = None

    @property
    def weight_tensors(self) -> List[torch.Tensor]:
        ts = []
        convs = [self.conv1] + [c for b in self.blocks
                                for c in (b.conv1, b.conv2, b.conv3, b.down) if c]
        for c in convs:
            ts.extend([c.wq, c.bias])
        ts.extend([self.fcq, self.fc_bias])
        return ts

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        """x: [N, 3, 224, 224] f32 (NCHW input, converted to NHWC inside).

        After capture_graph(), calls with the captured batch shape replay
        the recorded hipGraph (one launch for the whole ~120-kernel
        forward — wins where small batches are launch-bound)."""
        if self._graph is not None and x.shape == self._static_in.shape:
            self._static_in.copy_(x)
            self._graph.replay()
            return self._static_out
        return self._forward(x)

    def _forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x.permute(0, 2, 3, 1).contiguous()  # NHWC
        x = self.conv1(x)
        # maxpool 3x3 s2 p1: native NHWC kernel (no NCHW permute round trip)
        x = _ops.ext().maxpool2d_nhwc(x.contiguous(), 3, 2, 1)
        x = x.contiguous()
        for b in self.blocks:
            x = b(x)
        x = x.mean(dim=(1, 2))  # global average pool -> [N, 2048]
        e = _ops.ext()
        if self.fc_sa is None:
            self.fc_sa = _act_scale(x) * _CAL_HEADROOM
        sa = self.fc_sa
        logits = e.gemm_fp8(e.quantize_fp8(x.contiguous(), sa), self.fcq,
                            sa * self.fcscale, self.fc_bias, False)
        return logits

    def capture_graph(self, batch: int) -> None:
        """Record the forward as a hipGraph (torch.cuda.CUDAGraph on ROCm).

        Requires the calibration pass to have run (frozen activation
        scales: the captured graph must be sync-free). The whole fused
        forward replays as a single graph launch afterwards."""
        assert self.fc_sa is not None, "run one calibration batch first"
        self._static_in = torch.zeros(batch, 3, 224, 224,
                                      device=self.device)
        # warmup on a side stream (allocator state for capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._forward(self._static_in)
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._static_out = self._forward(self._static_in)
        self._graph = g


class LinearW4:
    """Serving Linear on the mixed MX GEMM: weights quantized ONCE to
    fp4-MX (half the resident bytes and half the weight traffic of fp8),
    activations quantized per call to fp8-MX by the 9-TB/s hw quantizer.
    Runs at the full fp8 MFMA rate (gemm_mx_w4: 1944/2293 TF @4096/8192).

    Requires serving-aligned shapes: batch % 256 == 0, out_features % 256
    == 0, in_features % 128 == 0 (the 8-phase scaled-MFMA tile).
    """

    def __init__(self, weight: torch.Tensor, bias: Optional[torch.Tensor] = None):
        # weight: [out_features, in_features] f32 on device
        assert weight.dim() == 2
        self.out_features, self.in_features = weight.shape
        assert self.out_features % 256 == 0 and self.in_features % 128 == 0, \
            "LinearW4 needs out%256==0 and in%128==0"
        e = _ops.ext()
        self.wq4, self.wsc4 = e.quantize_mx4(weight.contiguous())
        self.bias = bias

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        # x: [M, in_features] f32, M % 256 == 0
        e = _ops.ext()
        qa, sa = e.quantize_mx(x.contiguous())
        y = e.gemm_mx_w4(qa, sa, self.wq4, self.wsc4)
        if self.bias is not None:
            y = y + self.bias
        return y
