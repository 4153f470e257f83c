"""Gradient compression (reference: horovod/torch/compression.py:20-74).

The reference compresses in Python (tensor.half()) before the allreduce and
decompresses after — two extra full memory passes.  On MI355X the wire dtype
is handed to the native core instead and the conversion happens inside the
CDNA4 fusion pack/unpack kernels at zero extra HBM traffic.  The classic
compress()/decompress() methods are kept for API compatibility.
"""
import torch


class Compressor:
    """Interface: compress/decompress + the wire dtype the core should use."""

    @staticmethod
    def compress(tensor):
        return tensor, None

    @staticmethod
    def decompress(tensor, ctx):
        return tensor

    @staticmethod
    def wire_dtype(dtype):
        return None  # None = no conversion


class NoneCompressor(Compressor):
    pass


class FP16Compressor(Compressor):
    @staticmethod
    def compress(tensor):
        ctx = tensor.dtype
        if tensor.dtype.is_floating_point:
            tensor = tensor.type(torch.float16)
        return tensor, ctx

    @staticmethod
    def decompress(tensor, ctx):
        if ctx is not None and ctx.is_floating_point:
            tensor = tensor.type(ctx)
        return tensor

    @staticmethod
    def wire_dtype(dtype):
        if dtype in (torch.float32, torch.float64):
            return torch.float16
        return None


class BF16Compressor(Compressor):
    """MI355X-native addition: bf16 wire compression (same exponent range as
    fp32 — the safer choice for gradients on CDNA4)."""

    @staticmethod
    def compress(tensor):
        ctx = tensor.dtype
        if tensor.dtype.is_floating_point:
            tensor = tensor.type(torch.bfloat16)
        return tensor, ctx

    @staticmethod
    def decompress(tensor, ctx):
        if ctx is not None and ctx.is_floating_point:
            tensor = tensor.type(ctx)
        return tensor

    @staticmethod
    def wire_dtype(dtype):
        if dtype in (torch.float32, torch.float64):
            return torch.bfloat16
        return None


class Compression:
    none = NoneCompressor
    fp16 = FP16Compressor
    bf16 = BF16Compressor
