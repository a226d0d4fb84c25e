from .hourglass import (Mish, Activation, SPP, Pool, Convolution, Residual,
                        Hourglass, PreLayer, Neck, Head, StackedHourglass,
                        build_model)

__all__ = ['Mish', 'Activation', 'SPP', 'Pool', 'Convolution', 'Residual',
           'Hourglass', 'PreLayer', 'Neck', 'Head', 'StackedHourglass',
           'build_model']
