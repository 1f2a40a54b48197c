"""Fused ResNet bottleneck block + spatially-parallel variant.

API parity with the reference ``apex.contrib.bottleneck``
(apex/contrib/bottleneck/bottleneck.py: Bottleneck:154,
SpatialBottleneckFunction:304): frozen-BN folded into per-channel
scale/bias, conv+scale+bias+ReLU composed per branch, and a spatial variant
that splits H across a rank group and halo-exchanges ``dilation`` edge rows
before the 3x3 convolution (``spatial_method=1``; the output-halo variants
2/3 are a later round).

The reference drives cuDNN-frontend fusion; on MI355X the convs run through
MIOpen (torch conv2d) with the scale/bias/ReLU epilogue composed around them
(see conv_bias_relu.py for the fusion note).
"""

import torch

from ..conv_bias_relu import ConvFrozenScaleBiasReLU


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
        if self.spatial_method not in (0, 1):
            raise NotImplementedError(
                "spatial_method 2/3 (remote output-halo compute) lands in a later round"
            )

    def forward(self, x):
        if self.spatial_group_size <= 1:
            return super().forward(x)

        s1, b1 = self.bn1.get_scale_bias()
        s2, b2 = self.bn2.get_scale_bias()
        s3, b3 = self.bn3.get_scale_bias()

        out = ConvFrozenScaleBiasReLU(x, self.conv1.weight, s1, b1, 0, 1)

        # halo exchange: 1 row each side for the padded 3x3 conv
        top_halo = out[:, :, :1, :]
        btm_halo = out[:, :, -1:, :]
        left_in, right_in = self.spatial_halo_exchanger.left_right_halo_exchange(
            top_halo.contiguous(), btm_halo.contiguous()
        )
        parts = []
        if self.spatial_group_rank > 0:
            parts.append(left_in)
        parts.append(out)
        if self.spatial_group_rank < self.spatial_group_size - 1:
            parts.append(right_in)
        padded = torch.cat(parts, dim=2)

        mid = ConvFrozenScaleBiasReLU(padded, self.conv2.weight, s2, b2, 1, self.stride)
        # crop the rows produced by the halo padding
        top_crop = 1 if self.spatial_group_rank > 0 else 0
        h_local = out.shape[2] // self.stride if self.stride > 1 else out.shape[2]
        mid = mid[:, :, top_crop // max(self.stride, 1):, :]
        mid = mid[:, :, :h_local, :].contiguous()

        out3 = torch.nn.functional.conv2d(mid, self.conv3.weight) * s3 + b3
        if self.use_downsample:
            ds = self.downsample_bn(
                torch.nn.functional.conv2d(x, self.downsample_conv.weight, stride=self.stride)
            )
        else:
            ds = x
        return torch.relu(out3 + ds)
