from .autoencoder import Encoder, Decoder, EncoderOutput
from .quantizer import Quantizer
from .probclass import ProbClass
from .sinet import SiNet
from .sifinder import SiFinder
from .dsin import DSIN

__all__ = ["Encoder", "Decoder", "EncoderOutput", "Quantizer", "ProbClass",
           "SiNet", "SiFinder", "DSIN"]
