from .transformer import Transformer, Encoder, Decoder  # noqa: F401
