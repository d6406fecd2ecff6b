"""In-house speech-to-text model (SpeechToText feature).

The reference serves this feature through the FasterWhisper engine
(internal/modelcontroller/engine_fasterwhisper.go); here it is an in-house
Whisper-architecture model behind the same `/v1/audio/transcriptions`
surface: log-mel frontend (CPU, numpy) -> conv downsampling + transformer
encoder -> autoregressive decoder with cross-attention and incremental KV.

The transcription path is a one-shot batch job, not continuous batching:
an encoder pass then a short greedy decode (<=448 tokens), so it runs as
plain PyTorch modules — GEMM-bound work lands on hipBLASLt via rocm torch;
the paged-KV machinery of the LLM path would buy nothing here. Weights are
random-init under the synthetic tokenizer (no offline checkpoints in this
environment); `load_weights_whisper` maps HF safetensors names when a real
checkpoint directory is provided.
"""
from __future__ import annotations

import dataclasses
import math
from typing import Optional

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

SAMPLE_RATE = 16_000
N_FFT = 400
HOP = 160
CHUNK_SECONDS = 30


@dataclasses.dataclass
class SpeechConfig:
    n_mels: int = 80
    n_audio_ctx: int = 1500  # 30 s / 10 ms hop / conv stride 2
    n_text_ctx: int = 448
    n_state: int = 128
    n_head: int = 4
    n_audio_layer: int = 2
    n_text_layer: int = 2
    vocab_size: int = 2048
    sot_token: int = 1  # bos
    eot_token: int = 2  # eos


PRESETS = {
    # synthetic-scale preset for CPU tests and smoke
    "whisper-tiny": SpeechConfig(),
    # production-shaped preset (large-v3 dimensions)
    "whisper-large": SpeechConfig(
        n_mels=128, n_state=1280, n_head=20, n_audio_layer=32, n_text_layer=32,
        vocab_size=51_866,
    ),
}


# --------------------------------------------------------------- frontend
def _mel_filterbank(n_mels: int, n_fft: int = N_FFT, sr: int = SAMPLE_RATE) -> np.ndarray:
    """Slaney-style mel filterbank, computed analytically (no librosa)."""

    def hz_to_mel(f):
        return 2595.0 * np.log10(1.0 + np.asarray(f) / 700.0)

    def mel_to_hz(m):
        return 700.0 * (10.0 ** (np.asarray(m) / 2595.0) - 1.0)

    n_freqs = n_fft // 2 + 1
    freqs = np.linspace(0, sr / 2, n_freqs)
    mel_pts = mel_to_hz(np.linspace(hz_to_mel(0.0), hz_to_mel(sr / 2), n_mels + 2))
    fb = np.zeros((n_mels, n_freqs), dtype=np.float32)
    for i in range(n_mels):
        lo, ctr, hi = mel_pts[i], mel_pts[i + 1], mel_pts[i + 2]
        up = (freqs - lo) / max(ctr - lo, 1e-10)
        down = (hi - freqs) / max(hi - ctr, 1e-10)
        fb[i] = np.maximum(0.0, np.minimum(up, down))
    # Slaney area normalization
    enorm = 2.0 / (mel_pts[2 : n_mels + 2] - mel_pts[:n_mels])
    fb *= enorm[:, None].astype(np.float32)
    return fb


def log_mel_spectrogram(audio: np.ndarray, sr: int, n_mels: int = 80) -> np.ndarray:
    """waveform -> [n_mels, T] log-mel features (whisper-style scaling)."""
    audio = np.asarray(audio, dtype=np.float32)
    if audio.ndim > 1:
        audio = audio.mean(axis=-1)  # downmix
    if sr != SAMPLE_RATE:
        from scipy.signal import resample_poly

        g = math.gcd(int(sr), SAMPLE_RATE)
        audio = resample_poly(audio, SAMPLE_RATE // g, int(sr) // g).astype(np.float32)
    # pad/trim to one 30 s chunk
    target = CHUNK_SECONDS * SAMPLE_RATE
    if len(audio) < target:
        audio = np.pad(audio, (0, target - len(audio)))
    else:
        audio = audio[:target]
    window = np.hanning(N_FFT + 1)[:-1].astype(np.float32)
    # center-padded framing: exactly one frame per hop (target/HOP frames)
    padded = np.pad(audio, (N_FFT // 2, N_FFT // 2), mode="reflect")
    n_frames = target // HOP
    idx = np.arange(N_FFT)[None, :] + HOP * np.arange(n_frames)[:, None]
    frames = padded[idx] * window
    spec = np.abs(np.fft.rfft(frames, axis=-1)) ** 2  # [T, n_freq]
    mel = _mel_filterbank(n_mels) @ spec.T  # [n_mels, T]
    logmel = np.log10(np.maximum(mel, 1e-10))
    logmel = np.maximum(logmel, logmel.max() - 8.0)
    return ((logmel + 4.0) / 4.0).astype(np.float32)


# ----------------------------------------------------------------- model
class _MHA(nn.Module):
    def __init__(self, n_state: int, n_head: int):
        super().__init__()
        self.n_head = n_head
        self.q = nn.Linear(n_state, n_state)
        self.k = nn.Linear(n_state, n_state, bias=False)
        self.v = nn.Linear(n_state, n_state)
        self.out = nn.Linear(n_state, n_state)

    def forward(self, x, xa=None, kv_cache=None, causal=False):
        q = self.q(x)
        src = x if xa is None else xa
        if kv_cache is not None and xa is not None and self.k in kv_cache:
            k, v = kv_cache[self.k], kv_cache[self.v]  # cross-attn: static
        else:
            k, v = self.k(src), self.v(src)
            if kv_cache is not None:
                if xa is None:  # self-attn: append
                    if self.k in kv_cache:
                        k = torch.cat([kv_cache[self.k], k], dim=1)
                        v = torch.cat([kv_cache[self.v], v], dim=1)
                kv_cache[self.k], kv_cache[self.v] = k, v
        B, Tq, C = q.shape
        Tk = k.shape[1]
        h = self.n_head
        q = q.view(B, Tq, h, C // h).transpose(1, 2)
        k = k.view(B, Tk, h, C // h).transpose(1, 2)
        v = v.view(B, Tk, h, C // h).transpose(1, 2)
        o = F.scaled_dot_product_attention(q, k, v, is_causal=causal and Tq > 1)
        o = o.transpose(1, 2).reshape(B, Tq, C)
        return self.out(o)


class _Block(nn.Module):
    def __init__(self, n_state: int, n_head: int, cross: bool):
        super().__init__()
        self.attn_ln = nn.LayerNorm(n_state)
        self.attn = _MHA(n_state, n_head)
        self.cross_attn_ln = nn.LayerNorm(n_state) if cross else None
        self.cross_attn = _MHA(n_state, n_head) if cross else None
        self.mlp_ln = nn.LayerNorm(n_state)
        self.mlp = nn.Sequential(
            nn.Linear(n_state, 4 * n_state), nn.GELU(), nn.Linear(4 * n_state, n_state)
        )

    def forward(self, x, xa=None, kv_cache=None, causal=False):
        x = x + self.attn(self.attn_ln(x), kv_cache=kv_cache, causal=causal)
        if self.cross_attn is not None:
            x = x + self.cross_attn(self.cross_attn_ln(x), xa=xa, kv_cache=kv_cache)
        return x + self.mlp(self.mlp_ln(x))


def _sinusoids(length: int, channels: int) -> torch.Tensor:
    log_timescale = math.log(10000) / (channels // 2 - 1)
    inv = torch.exp(-log_timescale * torch.arange(channels // 2))
    t = torch.arange(length)[:, None] * inv[None, :]
    return torch.cat([t.sin(), t.cos()], dim=1)


class SpeechEncoder(nn.Module):
    def __init__(self, cfg: SpeechConfig):
        super().__init__()
        self.conv1 = nn.Conv1d(cfg.n_mels, cfg.n_state, 3, padding=1)
        self.conv2 = nn.Conv1d(cfg.n_state, cfg.n_state, 3, stride=2, padding=1)
        self.register_buffer(
            "pos", _sinusoids(cfg.n_audio_ctx, cfg.n_state), persistent=False
        )
        self.blocks = nn.ModuleList(
            _Block(cfg.n_state, cfg.n_head, cross=False)
            for _ in range(cfg.n_audio_layer)
        )
        self.ln_post = nn.LayerNorm(cfg.n_state)

    def forward(self, mel: torch.Tensor) -> torch.Tensor:  # [B, n_mels, T]
        x = F.gelu(self.conv1(mel))
        x = F.gelu(self.conv2(x)).permute(0, 2, 1)  # [B, T/2, C]
        x = x + self.pos[: x.shape[1]].to(x.dtype)
        for b in self.blocks:
            x = b(x)
        return self.ln_post(x)


class SpeechDecoder(nn.Module):
    def __init__(self, cfg: SpeechConfig):
        super().__init__()
        self.token_embedding = nn.Embedding(cfg.vocab_size, cfg.n_state)
        self.positional_embedding = nn.Parameter(
            torch.empty(cfg.n_text_ctx, cfg.n_state).normal_(std=0.02)
        )
        self.blocks = nn.ModuleList(
            _Block(cfg.n_state, cfg.n_head, cross=True)
            for _ in range(cfg.n_text_layer)
        )
        self.ln = nn.LayerNorm(cfg.n_state)

    def forward(self, tokens, xa, kv_caches, offset: int):
        x = (
            self.token_embedding(tokens)
            + self.positional_embedding[offset : offset + tokens.shape[1]]
        )
        for b, kc in zip(self.blocks, kv_caches):
            x = b(x, xa=xa, kv_cache=kc, causal=True)
        x = self.ln(x)
        return x @ self.token_embedding.weight.t()  # tied head


class SpeechToText(nn.Module):
    """End-to-end transcriber; greedy decode with incremental KV."""

    def __init__(self, cfg: SpeechConfig, device: str = "cpu", seed: int = 0):
        super().__init__()
        self.cfg = cfg
        torch.manual_seed(seed)
        self.encoder = SpeechEncoder(cfg)
        self.decoder = SpeechDecoder(cfg)
        self.to(device)
        self.device_ = device
        self.eval()

    @torch.inference_mode()
    def transcribe_tokens(
        self, audio: np.ndarray, sr: int, max_tokens: int = 64
    ) -> list[int]:
        cfg = self.cfg
        mel = torch.from_numpy(log_mel_spectrogram(audio, sr, cfg.n_mels))
        mel = mel[None].to(self.device_)
        xa = self.encoder(mel)
        kv_caches: list[dict] = [{} for _ in cfg.n_text_layer * [0]]
        toks = [cfg.sot_token]
        cur = torch.tensor([[cfg.sot_token]], device=self.device_)
        for i in range(min(max_tokens, cfg.n_text_ctx - 1)):
            logits = self.decoder(cur, xa, kv_caches, offset=i)
            row = logits[0, -1].clone()
            # the tied head makes hidden states correlate with the current
            # token's embedding; mask it (and sot) so greedy decoding cannot
            # collapse into a self-loop under random-init weights
            row[toks[-1]] = -torch.inf
            row[cfg.sot_token] = -torch.inf
            nxt = int(row.argmax())
            toks.append(nxt)
            if nxt == cfg.eot_token:
                break
            cur = torch.tensor([[nxt]], device=self.device_)
        return toks


# ----------------------------------------------------------- checkpoint IO
def save_whisper_checkpoint(model: "SpeechToText", out_dir: str) -> None:
    """Write the model in HF whisper layout (tests + weight-prep tooling)."""
    import os

    from safetensors.torch import save_file

    os.makedirs(out_dir, exist_ok=True)
    state = {}
    for name, p in model.state_dict().items():
        hf = (
            name.replace("encoder.", "model.encoder.")
            .replace("decoder.", "model.decoder.")
            .replace(".attn_ln.", ".self_attn_layer_norm.")
            .replace(".cross_attn_ln.", ".encoder_attn_layer_norm.")
            .replace(".mlp_ln.", ".final_layer_norm.")
            .replace(".attn.", ".self_attn.")
            .replace(".cross_attn.", ".encoder_attn.")
            .replace(".blocks.", ".layers.")
            .replace(".mlp.0.", ".fc1.").replace(".mlp.2.", ".fc2.")
            .replace(".q.", ".q_proj.").replace(".k.", ".k_proj.")
            .replace(".v.", ".v_proj.").replace(".out.", ".out_proj.")
            .replace(".token_embedding.", ".embed_tokens.")
            .replace("decoder.positional_embedding", "decoder.embed_positions.weight")
            .replace(".ln_post.", ".layer_norm.")
            .replace("decoder.ln.", "decoder.layer_norm.")
        )
        state[hf] = p.detach().cpu().contiguous()
    save_file(state, os.path.join(out_dir, "model.safetensors"))
    import json

    cfg = model.cfg
    with open(os.path.join(out_dir, "config.json"), "w") as f:
        json.dump({"architectures": ["WhisperForConditionalGeneration"],
                   "num_mel_bins": cfg.n_mels, "d_model": cfg.n_state,
                   "encoder_layers": cfg.n_audio_layer,
                   "decoder_layers": cfg.n_text_layer,
                   "encoder_attention_heads": cfg.n_head,
                   "decoder_attention_heads": cfg.n_head,
                   "max_source_positions": cfg.n_audio_ctx,
                   "max_target_positions": cfg.n_text_ctx,
                   "vocab_size": cfg.vocab_size,
                   "decoder_start_token_id": cfg.sot_token,
                   "eos_token_id": cfg.eot_token}, f)


def load_weights_whisper(model: "SpeechToText", model_dir: str) -> int:
    """Load an HF whisper checkpoint (model.encoder/decoder.* names) into
    SpeechToText; inverse of save_whisper_checkpoint. Returns params filled;
    raises if any engine parameter stays unset."""
    import glob
    import os

    from safetensors import safe_open

    inverse = {}
    for name in model.state_dict():
        hf = (
            name.replace("encoder.", "model.encoder.")
            .replace("decoder.", "model.decoder.")
            .replace(".attn_ln.", ".self_attn_layer_norm.")
            .replace(".cross_attn_ln.", ".encoder_attn_layer_norm.")
            .replace(".mlp_ln.", ".final_layer_norm.")
            .replace(".attn.", ".self_attn.")
            .replace(".cross_attn.", ".encoder_attn.")
            .replace(".blocks.", ".layers.")
            .replace(".mlp.0.", ".fc1.").replace(".mlp.2.", ".fc2.")
            .replace(".q.", ".q_proj.").replace(".k.", ".k_proj.")
            .replace(".v.", ".v_proj.").replace(".out.", ".out_proj.")
            .replace(".token_embedding.", ".embed_tokens.")
            .replace("decoder.positional_embedding", "decoder.embed_positions.weight")
            .replace(".ln_post.", ".layer_norm.")
            .replace("decoder.ln.", "decoder.layer_norm.")
        )
        inverse[hf] = name
    sd = model.state_dict()
    filled = set()
    files = sorted(glob.glob(os.path.join(model_dir, "*.safetensors")))
    if not files:
        raise FileNotFoundError(f"no *.safetensors under {model_dir}")
    for fpath in files:
        with safe_open(fpath, framework="pt", device="cpu") as sf:
            for key in sf.keys():
                tgt = inverse.get(key)
                if tgt is None:
                    continue  # proj_out etc. — decoder head is tied
                w = sf.get_tensor(key)
                with torch.no_grad():
                    sd[tgt].copy_(w.to(sd[tgt].dtype))
                filled.add(tgt)
    missing = set(sd) - filled
    if missing:
        raise ValueError(f"unfilled whisper parameters: {sorted(missing)[:8]}")
    return len(filled)
