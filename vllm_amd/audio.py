"""Audio input path: log-mel frontend + Whisper-style audio encoder.

Role of the reference's Whisper support (vllm/model_executor/models/
whisper.py encoder half + its feature extractor). The runner encodes a
request's waveform ONCE (cached on the request state, like vision
features); decoder layers cross-attend to the cached encoder states
(ForwardContext.cross_feats).

The mel frontend is torch-native (torch.stft + a hand-built triangular
filterbank), so it runs on CPU in tests and on the GPU in serving; the
encoder is plain PyTorch (prefill-sized, runs through rocBLAS GEMMs) —
conv1d x2 (stride-2 downsample) + sinusoidal positions + pre-norm
bidirectional transformer blocks, the Whisper encoder shape sized by
ModelSpec audio_* fields.
"""

from __future__ import annotations

import hashlib
import math

import torch
import torch.nn as nn
import torch.nn.functional as F

SAMPLE_RATE = 16_000
N_FFT = 400
HOP = 160


def mel_filterbank(n_mels: int, n_fft: int = N_FFT,
                   sr: int = SAMPLE_RATE) -> torch.Tensor:
    """Triangular mel filterbank [n_mels, n_fft//2+1] (Slaney-style
    spacing is not required for a from-scratch model; HTK mel scale)."""
    n_freqs = n_fft // 2 + 1
    freqs = torch.linspace(0, sr / 2, n_freqs)

    def hz_to_mel(f):
        return 2595.0 * math.log10(1.0 + f / 700.0)

    def mel_to_hz(m):
        return 700.0 * (10.0 ** (m / 2595.0) - 1.0)

    mel_pts = torch.linspace(hz_to_mel(0.0), hz_to_mel(sr / 2),
                             n_mels + 2)
    hz_pts = torch.tensor([mel_to_hz(float(m)) for m in mel_pts])
    fb = torch.zeros(n_mels, n_freqs)
    for i in range(n_mels):
        lo, ctr, hi = hz_pts[i], hz_pts[i + 1], hz_pts[i + 2]
        up = (freqs - lo) / torch.clamp(ctr - lo, min=1e-6)
        down = (hi - freqs) / torch.clamp(hi - ctr, min=1e-6)
        fb[i] = torch.clamp(torch.minimum(up, down), min=0.0)
    return fb


def log_mel_spectrogram(waveform: torch.Tensor,
                        n_mels: int) -> torch.Tensor:
    """waveform [S] float in [-1, 1] -> [n_mels, frames] normalized
    log-mel (Whisper's normalization: clamp to max-8 dB, /4 shift)."""
    device = getattr(waveform, "device", torch.device("cpu"))
    # The STFT runs on CPU regardless of input device: audio is a few
    # seconds of 1-D samples (microseconds of work) and this keeps the
    # hot path independent of the FFT backend on the GPU.
    waveform = torch.as_tensor(waveform, dtype=torch.float32)
    waveform = waveform.flatten().cpu()
    window = torch.hann_window(N_FFT)
    stft = torch.stft(waveform, N_FFT, HOP, window=window,
                      center=True, return_complex=True)
    power = stft.abs() ** 2  # [n_freqs, frames]
    mel = mel_filterbank(n_mels) @ power
    log = torch.clamp(mel, min=1e-10).log10()
    log = torch.maximum(log, log.max() - 8.0)
    return ((log + 4.0) / 4.0).to(device)


def _sinusoids(length: int, channels: int) -> torch.Tensor:
    """Whisper's fixed sinusoidal positions [length, channels]."""
    log_timescale = math.log(10_000) / (channels // 2 - 1)
    inv = torch.exp(-log_timescale * torch.arange(channels // 2))
    t = torch.arange(length).float().unsqueeze(1) * inv.unsqueeze(0)
    return torch.cat([t.sin(), t.cos()], dim=1)


class AudioEncoder(nn.Module):
    """Whisper-shaped audio encoder: mel -> conv downsample ->
    bidirectional transformer -> [frames/2, hidden]."""

    def __init__(self, n_mels: int, hidden: int, layers: int, heads: int,
                 max_frames: int, dtype: torch.dtype):
        super().__init__()
        self.n_mels = n_mels
        self.conv1 = nn.Conv1d(n_mels, hidden, 3, padding=1, dtype=dtype)
        self.conv2 = nn.Conv1d(hidden, hidden, 3, stride=2, padding=1,
                               dtype=dtype)
        self.register_buffer(
            "pos", _sinusoids(max_frames, hidden).to(dtype),
            persistent=False)
        self.blocks = nn.ModuleList([
            nn.ModuleDict({
                "ln1": nn.LayerNorm(hidden, dtype=dtype),
                "attn": nn.MultiheadAttention(hidden, heads,
                                              batch_first=True,
                                              dtype=dtype),
                "ln2": nn.LayerNorm(hidden, dtype=dtype),
                "fc1": nn.Linear(hidden, hidden * 4, dtype=dtype),
                "fc2": nn.Linear(hidden * 4, hidden, dtype=dtype),
            }) for _ in range(layers)
        ])
        self.post_ln = nn.LayerNorm(hidden, dtype=dtype)
        for p in self.parameters():
            p.requires_grad_(False)

    @torch.inference_mode()
    def forward(self, mel: torch.Tensor) -> torch.Tensor:
        """mel [n_mels, frames] -> [ceil(frames/2), hidden]."""
        x = F.gelu(self.conv1(mel.unsqueeze(0).to(self.pos.dtype)))
        x = F.gelu(self.conv2(x))
        x = x.transpose(1, 2)  # [1, T, H]
        t = x.shape[1]
        if t > self.pos.shape[0]:
            raise ValueError(
                f"audio too long: {t} frames > {self.pos.shape[0]}")
        x = x + self.pos[:t]
        for b in self.blocks:
            y = b["ln1"](x)
            x = x + b["attn"](y, y, y, need_weights=False)[0]
            y = b["ln2"](x)
            x = x + b["fc2"](F.gelu(b["fc1"](y)))
        return self.post_ln(x)[0]

    def init_dummy(self, seed: int) -> None:
        g = torch.Generator().manual_seed(seed ^ 0x61756469)  # 'audi'
        for p in self.parameters():
            with torch.no_grad():
                cpu = torch.empty(p.shape, dtype=torch.float32).normal_(
                    0.0, 0.25, generator=g)
                p.copy_(cpu.to(p.dtype))


def audio_content_hash(mm_data: dict) -> int:
    """Stable content hash of the waveform (block-hash salt: the same
    decoder prompt with different audio must not share KV blocks)."""
    wav = mm_data.get("audio")
    if wav is None:
        return 0
    t = torch.as_tensor(wav).float().cpu().contiguous()
    return int.from_bytes(
        hashlib.sha256(t.numpy().tobytes()).digest()[:8], "little")
