from .range_coder import (RangeEncoder, RangeDecoder, encode_with_freqs,
                          decode_with_freqs)
from .entropy import (ProbclassTesting, PredictionNetwork, encode_symbols,
                      decode_symbols)

__all__ = ["RangeEncoder", "RangeDecoder", "encode_with_freqs",
           "decode_with_freqs", "ProbclassTesting", "PredictionNetwork",
           "encode_symbols", "decode_symbols"]
