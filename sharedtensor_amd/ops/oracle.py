"""Reference ("oracle") implementations of the shared-tensor wire codecs.

These are the semantic ground truth for every native implementation in this
framework: the CPU C++ codec (csrc/codec_cpu.cpp) and the CDNA4 HIP kernels
(csrc/hip_kernels.hip) must match these bit-for-bit on the packed payload and
(for fp32 arithmetic) element-for-element on the residual update.

Wire-format parity with the reference implementation
(Hello1024/shared-tensor, /root/reference/src/sharedtensor.c):

* 1-bit codec: scale = 2^floor(log2(RMS(residual)))  (sharedtensor.c:156-159),
  one sign bit per element packed LSB-first into bytes (byte i/8, bit i%8,
  sharedtensor.c:166-174), bit==0 => +scale was sent, bit==1 => -scale
  (receiver applies values[i] += scale - bit*2*scale, sharedtensor.c:106-111).
  Exact error feedback: residual -= sent, in fp32 (sharedtensor.c:169-173).
  The packed byte payload here is byte-identical to the reference's
  (we accumulate the sum of squares in float64 rather than float32; the
  resulting power-of-two scale is the documented semantic).

* fp8 / int4 codecs are this framework's extension of the same contract
  (the reference README's TODO list asks for real delta compression kernels,
  /root/reference/README.md:47): per-packet power-of-two scale, quantized
  residual, exact fp32 error feedback (residual -= dequantized(sent)).

All functions are pure: they take a residual tensor and return
(scale(s), payload bytes, new_residual). Tensors are torch.float32 on CPU.
"""
from __future__ import annotations

import math
from typing import List, Sequence, Tuple

import numpy as np
import torch

CODEC_1BIT = 0
CODEC_FP8 = 1
CODEC_INT4 = 2

CODEC_NAMES = {"1bit": CODEC_1BIT, "fp8": CODEC_FP8, "int4": CODEC_INT4}
CODEC_IDS = {v: k for k, v in CODEC_NAMES.items()}

FP8_MAX = 448.0  # OCP e4m3fn max normal


def pow2_floor(x: float) -> float:
    """2^floor(log2(x)) for x > 0, else 0.0 (matches sharedtensor.c:159)."""
    if not (x > 0.0) or math.isinf(x) or math.isnan(x):
        return 0.0
    m, e = math.frexp(x)  # x = m * 2^e, m in [0.5, 1)
    return math.ldexp(1.0, e - 1)


def pow2_ceil(x: float) -> float:
    """2^ceil(log2(x)) for x > 0, else 0.0."""
    if not (x > 0.0) or math.isinf(x) or math.isnan(x):
        return 0.0
    m, e = math.frexp(x)
    if m == 0.5:  # exact power of two
        return math.ldexp(1.0, e - 1)
    return math.ldexp(1.0, e)


def rms_scale(delta: torch.Tensor) -> float:
    """Power-of-two RMS scale of a residual (sharedtensor.c:153-159).

    Sum of squares is accumulated in float64 (the reference uses float32
    accumulation; the pow2-floored result is what the protocol specifies).
    """
    assert delta.dtype == torch.float32
    ss = torch.sum(delta.double() * delta.double()).item()
    rms = math.sqrt(ss / max(delta.numel(), 1))
    return np.float32(pow2_floor(rms)).item()


# ---------------------------------------------------------------- 1-bit codec

def pad64(n: int) -> int:
    """Padded element count: each tensor's payload region covers a multiple
    of 64 elements so a 64-lane CDNA4 wavefront never straddles a tensor and
    __ballot packs exactly one uint64 word per wave."""
    return (n + 63) // 64 * 64


def words_1bit(n: int) -> int:
    """Payload size in uint64 words (8-byte aligned; superset of the
    reference's ceil(n/8) bytes — the first ceil(n/8) bytes are identical)."""
    return pad64(n) // 64


def encode_1bit(delta: torch.Tensor, scale: float = None) -> Tuple[float, bytes, torch.Tensor]:
    """Quantize+pack with error feedback (sharedtensor.c:166-174).

    residual > 0  -> bit 0, residual -= scale
    residual <= 0 -> bit 1, residual += scale
    Returns (scale, payload bytes (ceil(n/64)*8, LSB-first), new residual).
    """
    d = delta.detach().clone().float()
    n = d.numel()
    if scale is None:
        scale = rms_scale(d)
    scale = np.float32(scale).item()
    if scale == 0.0:
        return 0.0, b"\x00" * (words_1bit(n) * 8), d
    pos = d > 0
    bits = (~pos).view(-1).numpy().astype(np.uint8)
    sent = torch.where(pos, torch.full_like(d, scale), torch.full_like(d, -scale))
    new_d = d - sent  # exact fp32 error feedback
    packed = np.packbits(bits, bitorder="little")
    out = np.zeros(words_1bit(n) * 8, dtype=np.uint8)
    out[: packed.size] = packed
    return scale, out.tobytes(), new_d


def decode_1bit(payload: bytes, scale: float, n: int) -> torch.Tensor:
    """Return the per-element applied delta: +scale for bit 0, -scale for bit 1
    (sharedtensor.c:106-111)."""
    scale = np.float32(scale).item()
    if scale == 0.0:
        return torch.zeros(n, dtype=torch.float32)
    raw = np.frombuffer(payload, dtype=np.uint8)
    bits = np.unpackbits(raw, bitorder="little")[:n].astype(np.float32)
    return torch.from_numpy((1.0 - 2.0 * bits) * np.float32(scale)).float()


# ----------------------------------------------------------------- fp8 codec

def fp8_scale(delta: torch.Tensor) -> float:
    """Power-of-two scale so that max|residual|/scale <= FP8_MAX.

    Non-finite max |residual| -> 0.0 (keepalive semantics), matching
    cpu_compute_scale (codec_cpu.cpp) and k_finalize_scales: a poisoned
    residual is never quantized into the wire."""
    m = delta.abs().max().item()
    if m == 0.0 or math.isnan(m) or math.isinf(m):
        return 0.0
    return np.float32(pow2_ceil(m / FP8_MAX)).item()


def encode_fp8(delta: torch.Tensor, scale: float = None) -> Tuple[float, bytes, torch.Tensor]:
    """OCP e4m3fn quantization of residual/scale with exact error feedback.

    payload: one e4m3fn byte per element (n bytes, padded to 8-byte multiple).
    """
    d = delta.detach().clone().float()
    n = d.numel()
    if scale is None:
        scale = fp8_scale(d)
    scale = np.float32(scale).item()
    pad = pad64(n) - n
    if scale == 0.0:
        return 0.0, b"\x00" * (n + pad), d
    q = torch.clamp(d / scale, -FP8_MAX, FP8_MAX).to(torch.float8_e4m3fn)
    sent = q.float() * scale
    new_d = d - sent
    payload = q.view(torch.uint8).view(-1).numpy().tobytes() + b"\x00" * pad
    return scale, payload, new_d


def decode_fp8(payload: bytes, scale: float, n: int) -> torch.Tensor:
    scale = np.float32(scale).item()
    if scale == 0.0:
        return torch.zeros(n, dtype=torch.float32)
    raw = torch.frombuffer(bytearray(payload[:n]), dtype=torch.uint8)
    return raw.view(torch.float8_e4m3fn).float() * scale


# ---------------------------------------------------------------- int4 codec

def int4_scale(delta: torch.Tensor) -> float:
    """Power-of-two scale so that max|residual|/scale <= 7."""
    m = delta.abs().max().item()
    if m == 0.0 or math.isnan(m) or math.isinf(m):
        return 0.0
    return np.float32(pow2_ceil(m / 7.0)).item()


def encode_int4(delta: torch.Tensor, scale: float = None) -> Tuple[float, bytes, torch.Tensor]:
    """Symmetric int4 [-7, 7] quantization (round-to-nearest-even) with exact
    error feedback. payload: two's-complement nibbles, element i in byte i//2,
    low nibble for even i; ceil(n/2) bytes padded to 8-byte multiple."""
    d = delta.detach().clone().float()
    n = d.numel()
    if scale is None:
        scale = int4_scale(d)
    scale = np.float32(scale).item()
    nbytes = (n + 1) // 2
    pad = pad64(n) // 2 - nbytes
    if scale == 0.0:
        return 0.0, b"\x00" * (nbytes + pad), d
    q = torch.clamp(torch.round(d / scale), -7, 7).to(torch.int8)
    sent = q.float() * scale
    new_d = d - sent
    nib = (q.view(-1).numpy().astype(np.int8) & 0xF).astype(np.uint8)
    if n % 2:
        nib = np.concatenate([nib, np.zeros(1, dtype=np.uint8)])
    packed = (nib[0::2] | (nib[1::2] << 4)).astype(np.uint8)
    return scale, packed.tobytes() + b"\x00" * pad, new_d


def decode_int4(payload: bytes, scale: float, n: int) -> torch.Tensor:
    scale = np.float32(scale).item()
    if scale == 0.0:
        return torch.zeros(n, dtype=torch.float32)
    raw = np.frombuffer(payload, dtype=np.uint8)[: (n + 1) // 2]
    lo = (raw & 0xF).astype(np.int8)
    hi = (raw >> 4).astype(np.int8)
    # sign-extend 4-bit two's complement
    lo = np.where(lo > 7, lo - 16, lo)
    hi = np.where(hi > 7, hi - 16, hi)
    q = np.empty(raw.size * 2, dtype=np.int8)
    q[0::2] = lo
    q[1::2] = hi
    return torch.from_numpy(q[:n].astype(np.float32) * np.float32(scale))


# ------------------------------------------------------------ generic facade

def payload_bytes(codec: int, n: int) -> int:
    if codec == CODEC_1BIT:
        return pad64(n) // 8
    if codec == CODEC_FP8:
        return pad64(n)
    if codec == CODEC_INT4:
        return pad64(n) // 2
    raise ValueError(f"unknown codec {codec}")


def compute_scale(codec: int, delta: torch.Tensor) -> float:
    if codec == CODEC_1BIT:
        return rms_scale(delta)
    if codec == CODEC_FP8:
        return fp8_scale(delta)
    if codec == CODEC_INT4:
        return int4_scale(delta)
    raise ValueError(f"unknown codec {codec}")


def encode(codec: int, delta: torch.Tensor, scale: float = None):
    return {CODEC_1BIT: encode_1bit, CODEC_FP8: encode_fp8, CODEC_INT4: encode_int4}[codec](delta, scale)


def decode(codec: int, payload: bytes, scale: float, n: int) -> torch.Tensor:
    return {CODEC_1BIT: decode_1bit, CODEC_FP8: decode_fp8, CODEC_INT4: decode_int4}[codec](payload, scale, n)


# --------------------------------------------------- table (multi-tensor) form

def encode_table(codec: int, deltas: Sequence[torch.Tensor]) -> Tuple[List[float], bytes, List[torch.Tensor]]:
    """Per-tensor scales (reference README.md:41 'table sync ... per-tensor
    magnitude'), concatenated per-tensor payloads (each 8-byte aligned)."""
    scales, chunks, residuals = [], [], []
    for d in deltas:
        s, p, r = encode(codec, d)
        scales.append(s)
        chunks.append(p)
        residuals.append(r)
    return scales, b"".join(chunks), residuals


def decode_table(codec: int, payload: bytes, scales: Sequence[float], sizes: Sequence[int]) -> List[torch.Tensor]:
    out, off = [], 0
    for s, n in zip(scales, sizes):
        nb = payload_bytes(codec, n)
        out.append(decode(codec, payload[off: off + nb], s, n))
        off += nb
    return out
