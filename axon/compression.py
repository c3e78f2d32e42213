"""Gradient compression for the allreduce wire format.

Capability parity: horovod/torch/compression.py. The
reference casts fp32->fp16 in Python around the collective; here the cast
is fused into the CDNA4 pack/unpack kernels (kernels/pack_unpack.hip, K4):
a compressor contributes a `wire_dtype`, and the engine converts during the
fusion-buffer pack (one pass over the bytes instead of separate cast
kernels). The Python compress/decompress methods are kept for API
compatibility with user code that calls them directly.
"""
import torch


class Compressor:
    """Interface: wire_dtype drives the fused kernel path; compress /
    decompress implement the same transform at the Python level."""

    wire_dtype = None  # None = no compression

    @staticmethod
    def compress(tensor):
        return tensor, None

    @staticmethod
    def decompress(tensor, ctx):
        return tensor


class NoneCompressor(Compressor):
    wire_dtype = None


class FP16Compressor(Compressor):
    wire_dtype = torch.float16

    @staticmethod
    def compress(tensor):
        if tensor.dtype.is_floating_point:
            return tensor.to(torch.float16), tensor.dtype
        return tensor, None

    @staticmethod
    def decompress(tensor, ctx):
        return tensor.to(ctx) if ctx is not None else tensor


class BF16Compressor(Compressor):
    wire_dtype = torch.bfloat16

    @staticmethod
    def compress(tensor):
        if tensor.dtype.is_floating_point:
            return tensor.to(torch.bfloat16), tensor.dtype
        return tensor, None

    @staticmethod
    def decompress(tensor, ctx):
        return tensor.to(ctx) if ctx is not None else tensor


class Compression:
    """Namespace matching hvd.Compression.*"""

    none = NoneCompressor
    fp16 = FP16Compressor
    bf16 = BF16Compressor
