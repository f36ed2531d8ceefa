from .basic_layer import LinearLayer_Compress, QuantAct, fake_quantize
from .compress import init_compression, redundancy_clean

__all__ = ["init_compression", "redundancy_clean", "LinearLayer_Compress",
           "QuantAct", "fake_quantize"]
