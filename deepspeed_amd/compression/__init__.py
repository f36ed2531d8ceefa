from .basic_layer import LinearLayer_Compress, QuantAct, fake_quantize
from .compress import init_compression, redundancy_clean
from .distillation import (KDLoss, build_reduced_student,
                           CompressionScheduler)

__all__ = ["init_compression", "redundancy_clean", "LinearLayer_Compress",
           "QuantAct", "fake_quantize", "KDLoss", "build_reduced_student",
           "CompressionScheduler"]
