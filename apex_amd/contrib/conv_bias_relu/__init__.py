from .conv_bias_relu import ConvBias, ConvBiasReLU, ConvBiasMaskReLU, ConvFrozenScaleBiasReLU

__all__ = ["ConvBias", "ConvBiasReLU", "ConvBiasMaskReLU", "ConvFrozenScaleBiasReLU"]
