from .pointwise import linear_nd, add_gelu, gelu, linear_res_gelu
from .spectral import spectral_conv
from .projhead import proj_head, proj_head_supported
from .fft import rfft_trunc, fft_trunc, pad_ifft, pad_irfft
from .lifthead import lift_head, lift_head_supported
