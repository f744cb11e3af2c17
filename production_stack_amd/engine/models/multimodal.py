"""Engine-side vision + audio modalities (closes r1 VERDICT gap #40: the
router proxied /v1/audio/* and image content but no engine family served
them).

Design: the language model is unchanged; modalities are front-end encoders
whose output embeddings are INJECTED over placeholder token positions after
the embedding lookup (the llama-3.2-vision / whisper-style adapter
pattern, MI355X-first: encoders are bf16 torch modules whose GEMMs ride
hipBLASLt; the LM path keeps its HIP kernels).

* VisionEncoder: 32px patchify -> ViT blocks -> projector to the LM's
  hidden size. One 224x224 image -> 49 embedding tokens.
* AudioEncoder: log-mel (torch STFT) -> 2x strided conv -> transformer
  blocks -> projector. ~50 tokens per 10 s of 16 kHz audio.

Placeholder prompt tokens are derived from a BLAKE2 hash of the media
bytes, so prefix caching distinguishes different media at the same prompt
position for free, and identical media reuse cached KV.

Weights are random-init (no network in this environment), matching the
text families' north-star setup; everything is deterministic per seed.
"""

from __future__ import annotations

import hashlib
import io
import math
from typing import List, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


class _Block(nn.Module):
    def __init__(self, width: int, heads: int = 4) -> None:
        super().__init__()
        self.norm1 = nn.LayerNorm(width, dtype=torch.bfloat16)
        self.qkv = nn.Linear(width, 3 * width, dtype=torch.bfloat16)
        self.proj = nn.Linear(width, width, dtype=torch.bfloat16)
        self.norm2 = nn.LayerNorm(width, dtype=torch.bfloat16)
        self.fc1 = nn.Linear(width, 4 * width, dtype=torch.bfloat16)
        self.fc2 = nn.Linear(4 * width, width, dtype=torch.bfloat16)
        self.heads = heads
        self.width = width

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        T, W = x.shape
        h = self.norm1(x)
        qkv = self.qkv(h).view(T, 3, self.heads, W // self.heads)
        q, k, v = qkv.unbind(1)
        att = F.scaled_dot_product_attention(
            q.transpose(0, 1), k.transpose(0, 1), v.transpose(0, 1)
        )
        x = x + self.proj(att.transpose(0, 1).reshape(T, W))
        x = x + self.fc2(F.gelu(self.fc1(self.norm2(x))))
        return x


class VisionEncoder(nn.Module):
    IMG = 224
    PATCH = 32

    def __init__(self, hidden_size: int, width: int = 256,
                 depth: int = 2, seed: int = 1234) -> None:
        super().__init__()
        n_patch = (self.IMG // self.PATCH) ** 2  # 49
        self.patch = nn.Linear(3 * self.PATCH * self.PATCH, width,
                               dtype=torch.bfloat16)
        self.pos = nn.Parameter(
            torch.empty(n_patch, width, dtype=torch.bfloat16))
        self.blocks = nn.ModuleList(_Block(width) for _ in range(depth))
        self.out = nn.Linear(width, hidden_size, dtype=torch.bfloat16)
        g = torch.Generator().manual_seed(seed)
        with torch.no_grad():
            for p in self.parameters():
                p.copy_(torch.randn(p.shape, generator=g,
                                    dtype=torch.float32)
                        .to(torch.bfloat16) * 0.02)

    @torch.no_grad()
    def forward(self, img: torch.Tensor) -> torch.Tensor:
        """img [3, 224, 224] float in [0,1] -> [49, hidden]"""
        P = self.PATCH
        x = img.unfold(1, P, P).unfold(2, P, P)  # [3, 7, 7, P, P]
        x = x.permute(1, 2, 0, 3, 4).reshape(-1, 3 * P * P)
        x = self.patch(x.to(torch.bfloat16)) + self.pos
        for b in self.blocks:
            x = b(x)
        return self.out(x)


class AudioEncoder(nn.Module):
    SR = 16000
    N_MELS = 64
    MAX_FRAMES = 1500

    def __init__(self, hidden_size: int, width: int = 256,
                 depth: int = 2, seed: int = 4321) -> None:
        super().__init__()
        self.conv1 = nn.Conv1d(self.N_MELS, width, 3, stride=2, padding=1,
                               dtype=torch.bfloat16)
        self.conv2 = nn.Conv1d(width, width, 3, stride=2, padding=1,
                               dtype=torch.bfloat16)
        self.blocks = nn.ModuleList(_Block(width) for _ in range(depth))
        self.out = nn.Linear(width, hidden_size, dtype=torch.bfloat16)
        g = torch.Generator().manual_seed(seed)
        with torch.no_grad():
            for p in self.parameters():
                p.copy_(torch.randn(p.shape, generator=g,
                                    dtype=torch.float32)
                        .to(torch.bfloat16) * 0.02)
        # mel filterbank (triangular, HTK-style spacing)
        self.register_buffer("mel_fb", _mel_filterbank(self.N_MELS, 400,
                                                       self.SR),
                             persistent=False)

    @torch.no_grad()
    def forward(self, wav: torch.Tensor) -> torch.Tensor:
        """wav [T] float32 (16 kHz mono) -> [frames//16, hidden]"""
        spec = torch.stft(wav, n_fft=400, hop_length=160,
                          window=torch.hann_window(400, device=wav.device),
                          return_complex=True).abs() ** 2
        mel = (self.mel_fb.to(wav.device) @ spec).clamp(min=1e-8).log()
        mel = mel[:, : self.MAX_FRAMES]
        x = F.gelu(self.conv1(mel.to(torch.bfloat16).unsqueeze(0)))
        x = F.gelu(self.conv2(x))[0].transpose(0, 1)  # [frames/4, width]
        x = x[::4]  # 1 token per ~40 ms
        for b in self.blocks:
            x = b(x)
        return self.out(x)


def _mel_filterbank(n_mels: int, n_fft: int, sr: int) -> torch.Tensor:
    def hz_to_mel(f):
        return 2595.0 * math.log10(1.0 + f / 700.0)

    def mel_to_hz(m):
        return 700.0 * (10.0 ** (m / 2595.0) - 1.0)

    n_bins = n_fft // 2 + 1
    mels = torch.linspace(hz_to_mel(0), hz_to_mel(sr / 2), n_mels + 2)
    hz = torch.tensor([mel_to_hz(m.item()) for m in mels])
    bins = (hz / (sr / 2) * (n_bins - 1)).long()
    fb = torch.zeros(n_mels, n_bins)
    for i in range(n_mels):
        lo, mid, hi = bins[i], bins[i + 1], bins[i + 2]
        if mid > lo:
            fb[i, lo:mid] = torch.linspace(0, 1, int(mid - lo))
        if hi > mid:
            fb[i, mid:hi] = torch.linspace(1, 0, int(hi - mid))
    return fb


# ---------------------------------------------------------------------------
def decode_image(data: bytes) -> torch.Tensor:
    """image bytes (PNG/JPEG/BMP/PPM via PIL) -> [3, 224, 224] float."""
    from PIL import Image

    img = Image.open(io.BytesIO(data)).convert("RGB").resize(
        (VisionEncoder.IMG, VisionEncoder.IMG))
    arr = torch.frombuffer(bytearray(img.tobytes()), dtype=torch.uint8)
    arr = arr.view(VisionEncoder.IMG, VisionEncoder.IMG, 3)
    return arr.permute(2, 0, 1).float() / 255.0


def decode_wav(data: bytes) -> torch.Tensor:
    """WAV bytes -> [T] float32 mono at the encoder's sample rate."""
    import wave as wave_mod

    with wave_mod.open(io.BytesIO(data)) as w:
        n = w.getnframes()
        raw = w.readframes(n)
        ch = w.getnchannels()
        sw = w.getsampwidth()
        sr = w.getframerate()
    if sw == 2:
        x = torch.frombuffer(bytearray(raw), dtype=torch.int16).float()
        x = x / 32768.0
    elif sw == 1:
        x = torch.frombuffer(bytearray(raw), dtype=torch.uint8).float()
        x = (x - 128.0) / 128.0
    else:
        x = torch.frombuffer(bytearray(raw), dtype=torch.int32).float()
        x = x / 2147483648.0
    if ch > 1:
        x = x.view(-1, ch).mean(dim=1)
    if sr != AudioEncoder.SR and sr > 0:
        idx = torch.linspace(0, x.numel() - 1,
                             int(x.numel() * AudioEncoder.SR / sr))
        x = x[idx.long()]
    return x


def media_placeholder_tokens(data: bytes, n: int, vocab: int,
                             reserved: int = 16) -> List[int]:
    """Deterministic pseudo-token ids from the media content hash: prefix
    caching then keys identical media to identical blocks and distinct
    media to distinct blocks."""
    out: List[int] = []
    counter = 0
    while len(out) < n:
        h = hashlib.blake2b(data + counter.to_bytes(4, "little"),
                            digest_size=32).digest()
        for i in range(0, 32, 4):
            if len(out) >= n:
                break
            v = int.from_bytes(h[i:i + 4], "little")
            out.append(reserved + v % (vocab - reserved))
        counter += 1
    return out
