"""Engine-side vision + audio modalities (closes r1 gap #40: router
proxied /v1/audio/* and image parts with no engine family behind them)."""

import base64
import io
import struct
import wave

import numpy as np
import pytest
import torch

from production_stack_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    SchedulerConfig,
)
from production_stack_amd.engine.engine import LLMEngine
from production_stack_amd.engine.sampling import SamplingParams


def tiny_engine():
    cfg = EngineConfig(
        model="tiny-llama", max_model_len=512, seed=11,
        cache=CacheConfig(block_size=16, num_gpu_blocks=256,
                          enable_prefix_caching=True),
        scheduler=SchedulerConfig(max_num_seqs=8,
                                  max_num_batched_tokens=256),
    )
    return LLMEngine(cfg, device="cpu")


def png_bytes(color):
    from PIL import Image

    img = Image.new("RGB", (64, 64), color)
    buf = io.BytesIO()
    img.save(buf, format="PNG")
    return buf.getvalue()


def wav_bytes(freq=440.0, seconds=0.5, sr=16000):
    t = np.arange(int(sr * seconds)) / sr
    x = (np.sin(2 * np.pi * freq * t) * 20000).astype(np.int16)
    buf = io.BytesIO()
    with wave.open(buf, "wb") as w:
        w.setnchannels(1)
        w.setsampwidth(2)
        w.setframerate(sr)
        w.writeframes(x.tobytes())
    return buf.getvalue()


def test_vision_injection_changes_output():
    """Same text prompt + different images -> different generations;
    same image -> identical generation (deterministic encoders +
    content-hash placeholders)."""
    from production_stack_amd.engine.models.multimodal import (
        decode_image,
        media_placeholder_tokens,
    )

    eng = tiny_engine()
    enc = eng.get_vision_encoder()

    def run(img_bytes):
        emb = enc(decode_image(img_bytes))
        toks = [5, 6, 7] + media_placeholder_tokens(
            img_bytes, emb.shape[0], eng.model_cfg.vocab_size) + [9, 10]
        rid = f"r{hash(img_bytes) & 0xffff}-{np.random.randint(1e9)}"
        eng.add_request(rid, toks, SamplingParams(
            max_tokens=8, temperature=0.0, ignore_eos=True),
            mm_embeds=[(3, emb)])
        outs = []
        while eng.has_unfinished():
            for o in eng.step():
                outs.extend(o.new_token_ids)
        return outs

    red1 = run(png_bytes((255, 0, 0)))
    red2 = run(png_bytes((255, 0, 0)))
    blue = run(png_bytes((0, 0, 255)))
    assert red1 == red2
    assert red1 != blue


def test_vision_injection_vs_no_injection():
    """Placeholder ids with injected embeddings generate differently than
    the same ids without injection (the hidden states really change)."""
    eng = tiny_engine()
    from production_stack_amd.engine.models.multimodal import (
        decode_image, media_placeholder_tokens,
    )
    data = png_bytes((0, 200, 0))
    emb = eng.get_vision_encoder()(decode_image(data))
    toks = [5] + media_placeholder_tokens(
        data, emb.shape[0], eng.model_cfg.vocab_size) + [9]
    p = SamplingParams(max_tokens=8, temperature=0.0, ignore_eos=True)
    eng.add_request("with", toks, p, mm_embeds=[(1, emb)])
    eng.add_request("without", toks, p)
    outs = {"with": [], "without": []}
    while eng.has_unfinished():
        for o in eng.step():
            outs[o.request_id].extend(o.new_token_ids)
    assert outs["with"] != outs["without"]


def test_audio_encoder_shapes_and_decode():
    from production_stack_amd.engine.models.multimodal import decode_wav

    eng = tiny_engine()
    wav = decode_wav(wav_bytes())
    assert wav.ndim == 1 and wav.numel() == 8000
    emb = eng.get_audio_encoder()(wav)
    assert emb.shape[1] == eng.model_cfg.hidden_size
    assert 2 <= emb.shape[0] <= 64


def test_http_chat_image_and_audio_endpoints():
    """Full HTTP round trip: chat with an image content part and
    /v1/audio/transcriptions (multipart + raw body)."""
    from tests.test_full_stack_cpu import RealEngineServer

    server = RealEngineServer(18790)
    server.start()
    try:
        import requests

        url = server.url
        img64 = base64.b64encode(png_bytes((10, 30, 200))).decode()
        r = requests.post(url + "/v1/chat/completions", json={
            "model": "tiny-llama",
            "messages": [{"role": "user", "content": [
                {"type": "text", "text": "describe"},
                {"type": "image_url", "image_url": {
                    "url": f"data:image/png;base64,{img64}"}},
            ]}],
            "max_tokens": 6,
        }, timeout=60)
        assert r.status_code == 200, r.text
        assert r.json()["choices"][0]["message"]["content"]

        wav = wav_bytes()
        boundary = "XBOUNDARY"
        body = (
            f"--{boundary}\r\nContent-Disposition: form-data; "
            f'name="file"; filename="a.wav"\r\n'
            f"Content-Type: audio/wav\r\n\r\n"
        ).encode() + wav + (
            f"\r\n--{boundary}\r\nContent-Disposition: form-data; "
            f'name="response_format"\r\n\r\njson\r\n--{boundary}--\r\n'
        ).encode()
        r = requests.post(
            url + "/v1/audio/transcriptions", data=body,
            headers={"Content-Type":
                     f"multipart/form-data; boundary={boundary}"},
            timeout=60,
        )
        assert r.status_code == 200, r.text
        assert isinstance(r.json().get("text"), str)

        # raw-body variant (translations alias)
        r = requests.post(url + "/v1/audio/translations", data=wav,
                          headers={"Content-Type": "audio/wav"},
                          timeout=60)
        assert r.status_code == 200, r.text
    finally:
        server.stop()
