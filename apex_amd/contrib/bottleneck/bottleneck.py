"""Fused ResNet bottleneck block + spatially-parallel variant.

API parity with the reference ``apex.contrib.bottleneck``
(apex/contrib/bottleneck/bottleneck.py: Bottleneck:154,
SpatialBottleneckFunction:304): frozen-BN folded into per-channel
scale/bias, conv+scale+bias+ReLU composed per branch, and a spatial variant
that splits H across a rank group: ``spatial_method=1`` splices exchanged
input halos and runs one conv; methods 2/3 run the interior conv on the
unpadded local slab and recompute only the two edge output rows from the
received halos (3 overlaps those small convs on a side stream).

The reference drives cuDNN-frontend fusion; on MI355X the convs run through
MIOpen (torch conv2d) with the scale/bias/ReLU epilogue composed around them
(see conv_bias_relu.py for the fusion note).
"""

import contextlib

import torch

from ..conv_bias_relu import ConvFrozenScaleBiasReLU


def _null_ctx():
    return contextlib.nullcontext()


class FrozenBatchNorm2d(torch.nn.Module):
    """BatchNorm2d with fixed statistics and affine parameters (the
    reference's norm_func for detection backbones)."""

    def __init__(self, num_features):
        super().__init__()
        self.register_buffer("weight", torch.ones(num_features))
        self.register_buffer("bias", torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))

    def get_scale_bias(self, nhwc=False):
        scale = self.weight * self.running_var.rsqrt()
        bias = self.bias - self.running_mean * scale
        if nhwc:
            return scale.reshape(1, 1, 1, -1), bias.reshape(1, 1, 1, -1)
        return scale.reshape(1, -1, 1, 1), bias.reshape(1, -1, 1, 1)

    def forward(self, x):
        scale, bias = self.get_scale_bias()
        return x * scale + bias


def kaiming_uniform_(tensor, a=1):
    return torch.nn.init.kaiming_uniform_(tensor, a=a)


class Bottleneck(torch.nn.Module):
    """1x1 -> 3x3 -> 1x1 bottleneck with frozen-BN scale/bias folding."""

    def __init__(self, in_channels, bottleneck_channels, out_channels, stride=1, groups=1,
                 dilation=1, norm_func=FrozenBatchNorm2d, use_cudnn=False, explicit_nhwc=False):
        super().__init__()
        if groups != 1:
            raise RuntimeError("Only support groups == 1")
        if dilation != 1:
            raise RuntimeError("Only support dilation == 1")
        self.stride = stride
        self.explicit_nhwc = explicit_nhwc

        self.conv1 = torch.nn.Conv2d(in_channels, bottleneck_channels, 1, bias=False)
        self.conv2 = torch.nn.Conv2d(bottleneck_channels, bottleneck_channels, 3, stride=stride,
                                     padding=1, bias=False)
        self.conv3 = torch.nn.Conv2d(bottleneck_channels, out_channels, 1, bias=False)
        self.bn1 = norm_func(bottleneck_channels)
        self.bn2 = norm_func(bottleneck_channels)
        self.bn3 = norm_func(out_channels)

        self.use_downsample = in_channels != out_channels or stride != 1
        if self.use_downsample:
            self.downsample_conv = torch.nn.Conv2d(in_channels, out_channels, 1, stride=stride,
                                                   bias=False)
            self.downsample_bn = norm_func(out_channels)
        for c in [self.conv1, self.conv2, self.conv3]:
            kaiming_uniform_(c.weight, a=1)

    def forward(self, x):
        s1, b1 = self.bn1.get_scale_bias()
        s2, b2 = self.bn2.get_scale_bias()
        s3, b3 = self.bn3.get_scale_bias()

        out = ConvFrozenScaleBiasReLU(x, self.conv1.weight, s1, b1, 0, 1)
        out = ConvFrozenScaleBiasReLU(out, self.conv2.weight, s2, b2, 1, self.stride)
        out = torch.nn.functional.conv2d(out, self.conv3.weight) * s3 + b3
        if self.use_downsample:
            ds = self.downsample_bn(
                torch.nn.functional.conv2d(x, self.downsample_conv.weight, stride=self.stride)
            )
        else:
            ds = x
        return torch.relu(out + ds)


class SpatialBottleneck(Bottleneck):
    """Bottleneck with the middle 3x3 conv split over H across a rank group;
    edge rows are halo-exchanged before the conv (spatial_method=1)."""

    def __init__(self, in_channels, bottleneck_channels, out_channels, stride=1, groups=1,
                 dilation=1, norm_func=FrozenBatchNorm2d, use_cudnn=False, explicit_nhwc=False,
                 spatial_parallel_args=None):
        super().__init__(in_channels, bottleneck_channels, out_channels, stride, groups,
                         dilation, norm_func, use_cudnn, explicit_nhwc)
        if spatial_parallel_args is None:
            self.spatial_args = (1, 0, None, None, 0, False)
        else:
            assert len(spatial_parallel_args) == 6, "spatial_parallel_args must have 6 elements"
            self.spatial_args = spatial_parallel_args
        (self.spatial_group_size, self.spatial_group_rank, self.spatial_communicator,
         self.spatial_halo_exchanger, self.spatial_method, self.use_delay_kernel) = self.spatial_args
        if self.spatial_method not in (0, 1, 2, 3):
            raise ValueError(f"unknown spatial_method {self.spatial_method}")

    def _conv2_input_halo(self, out, s2, b2):
        """spatial_method 1: splice exchanged input halos (or the global
        zero-padding rows at the outer boundaries) onto the local slab, run
        one conv with explicit rows instead of vertical padding, crop.

        Using explicit boundary rows keeps the stride phase right: the
        unpadded 3x3 conv's first output centers on slab row 1, which is
        always the rank's first global center (r * h_local is a multiple of
        the stride because h_local is)."""
        gr, gs = self.spatial_group_rank, self.spatial_group_size
        top_halo = out[:, :, :1, :]
        btm_halo = out[:, :, -1:, :]
        left_in, right_in = self.spatial_halo_exchanger.left_right_halo_exchange(
            top_halo.contiguous(), btm_halo.contiguous()
        )
        zero_row = torch.zeros_like(out[:, :, :1, :])
        parts = [left_in if gr > 0 else zero_row, out,
                 right_in if gr < gs - 1 else zero_row]
        slab = torch.cat(parts, dim=2)
        mid = ConvFrozenScaleBiasReLU(slab, self.conv2.weight, s2, b2, (0, 1), self.stride)
        h_out_local = out.shape[2] // self.stride
        return mid[:, :, :h_out_local, :].contiguous()

    def _conv2_edge_correction(self, out, s2, b2):
        """spatial_method 2/3: the interior conv runs on the local slab only
        (zero-padded) while halos are in flight; the two edge output rows are
        then recomputed from the received halo rows and spliced in
        (reference: output-halo compute, bottleneck.py spatial_method 2/3).
        Method 3 additionally runs the small edge convs on the exchanger's
        side stream so they overlap the 1x1 conv3 launch on the main stream.
        """
        gr, gs, s = self.spatial_group_rank, self.spatial_group_size, self.stride
        top_halo = out[:, :, :1, :]
        btm_halo = out[:, :, -1:, :]
        left_in, right_in = self.spatial_halo_exchanger.left_right_halo_exchange(
            top_halo.contiguous(), btm_halo.contiguous()
        )
        # interior: the zero-padded local conv; output centers land on local
        # rows 0, s, 2s, … which are globally phase-aligned (h_local % s == 0)
        mid = ConvFrozenScaleBiasReLU(out, self.conv2.weight, s2, b2, 1, s)

        def edge(rows3):
            # 3 input rows -> 1 output row; W keeps the conv's stride
            z = torch.nn.functional.conv2d(rows3, self.conv2.weight,
                                           stride=(1, s), padding=(0, 1))
            return torch.relu(z * s2 + b2)

        # which output rows have a receptive field reaching a neighbour:
        # the first (center local row 0, needs the left halo) for every
        # stride; the last (center h_local-1) only when s == 1
        need_top = gr > 0
        need_btm = s == 1 and gr < gs - 1
        stream = getattr(self.spatial_halo_exchanger, "stream1", None)
        use_side = self.spatial_method == 3 and stream is not None and out.is_cuda
        ctx = torch.cuda.stream(stream) if use_side else _null_ctx()
        if use_side:
            stream.wait_stream(torch.cuda.current_stream())
        with ctx:
            top_row = edge(torch.cat([left_in, out[:, :, :2, :]], dim=2)) if need_top else None
            btm_row = edge(torch.cat([out[:, :, -2:, :], right_in], dim=2)) if need_btm else None
        if use_side:
            torch.cuda.current_stream().wait_stream(stream)
        rows = [mid[:, :, 1:, :] if need_top else mid]
        if need_top:
            rows.insert(0, top_row)
        if need_btm:
            rows[-1] = rows[-1][:, :, :-1, :]
            rows.append(btm_row)
        return torch.cat(rows, dim=2)

    def forward(self, x):
        if self.spatial_group_size <= 1:
            return super().forward(x)

        s1, b1 = self.bn1.get_scale_bias()
        s2, b2 = self.bn2.get_scale_bias()
        s3, b3 = self.bn3.get_scale_bias()

        out = ConvFrozenScaleBiasReLU(x, self.conv1.weight, s1, b1, 0, 1)

        if self.spatial_method in (2, 3):
            mid = self._conv2_edge_correction(out, s2, b2)
        else:  # methods 0/1
            mid = self._conv2_input_halo(out, s2, b2)

        out3 = torch.nn.functional.conv2d(mid, self.conv3.weight) * s3 + b3
        if self.use_downsample:
            ds = self.downsample_bn(
                torch.nn.functional.conv2d(x, self.downsample_conv.weight, stride=self.stride)
            )
        else:
            ds = x
        return torch.relu(out3 + ds)
