"""CPU validation of the reference op implementations themselves.

The paged references are checked against dense (non-paged) attention so the
GPU kernel tests compare against a trustworthy baseline.
"""

import torch

from production_stack_amd.ops import reference


def test_rms_norm_matches_manual():
    x = torch.randn(5, 64, dtype=torch.bfloat16)
    w = torch.randn(64, dtype=torch.bfloat16)
    out = reference.rms_norm(x, w, 1e-6)
    xf = x.float()
    want = xf / (xf.pow(2).mean(-1, keepdim=True) + 1e-6).sqrt() * w.float()
    torch.testing.assert_close(out.float(), want, atol=2e-2, rtol=2e-2)


def test_paged_decode_matches_dense_attention():
    torch.manual_seed(0)
    S, qh, kh, hd, bs = 3, 8, 2, 64, 16
    seq_lens = torch.tensor([5, 33, 16], dtype=torch.int32)
    max_blocks = 4
    nb = S * max_blocks + 1
    k_cache = torch.randn(nb, kh, bs, hd, dtype=torch.bfloat16)
    v_cache = torch.randn(nb, kh, bs, hd, dtype=torch.bfloat16)
    bt = torch.arange(1, S * max_blocks + 1, dtype=torch.int32).reshape(S, max_blocks)
    q = torch.randn(S, qh, hd, dtype=torch.bfloat16)
    out = reference.paged_attn_decode(q, k_cache, v_cache, bt, seq_lens, 0.125)

    for s in range(S):
        ctx = int(seq_lens[s])
        k = reference._gather_kv(k_cache, bt[s], ctx).float()
        v = reference._gather_kv(v_cache, bt[s], ctx).float()
        for h in range(qh):
            kvh = h // (qh // kh)
            attn = torch.softmax((k[:, kvh] @ q[s, h].float()) * 0.125, dim=-1)
            want = attn @ v[:, kvh]
            torch.testing.assert_close(
                out[s, h].float(), want, atol=2e-2, rtol=2e-2
            )


def test_prefill_is_causal():
    """A prefill token at position p must ignore cache entries > p."""
    torch.manual_seed(0)
    qh, kh, hd, bs = 2, 2, 64, 16
    nb, max_blocks = 5, 4
    k_cache = torch.randn(nb, kh, bs, hd, dtype=torch.bfloat16)
    v_cache = torch.randn(nb, kh, bs, hd, dtype=torch.bfloat16)
    bt = torch.arange(1, 5, dtype=torch.int32).reshape(1, max_blocks)
    q = torch.randn(2, qh, hd, dtype=torch.bfloat16)
    token_seq = torch.zeros(2, dtype=torch.int32)
    token_pos = torch.tensor([3, 7], dtype=torch.int32)
    out = reference.paged_attn_prefill(
        q, k_cache, v_cache, bt, token_seq, token_pos, 0.125
    )
    # poisoning cache entries beyond each token's position must not change it
    k2, v2 = k_cache.clone(), v_cache.clone()
    k2[bt[0, 1].long(), :, :] = 1e4  # tokens 16.. (beyond pos 7)
    v2[bt[0, 1].long(), :, :] = -1e4
    out2 = reference.paged_attn_prefill(
        q, k2, v2, bt, token_seq, token_pos, 0.125
    )
    torch.testing.assert_close(out, out2)


def test_rope_preserves_tail_dims():
    """rot_dim < head_dim leaves the tail dims untouched."""
    T, qh, kh, hd, rot = 4, 2, 1, 64, 32
    pos = torch.arange(T, dtype=torch.int32)
    inv = 1.0 / (10000.0 ** (torch.arange(0, rot, 2).float() / rot))
    freqs = torch.outer(torch.arange(16).float(), inv)
    cos_sin = torch.cat([freqs.cos(), freqs.sin()], -1)
    q = torch.randn(T, qh * hd, dtype=torch.bfloat16)
    k = torch.randn(T, kh * hd, dtype=torch.bfloat16)
    q2, k2 = reference.rotary_embedding(pos, q, k, cos_sin, hd)
    torch.testing.assert_close(
        q2.view(T, qh, hd)[..., rot:], q.view(T, qh, hd)[..., rot:]
    )
    # position 0 is the identity rotation
    torch.testing.assert_close(
        q2.view(T, qh, hd)[0, :, :rot].float(),
        q.view(T, qh, hd)[0, :, :rot].float(),
        atol=1e-2,
        rtol=1e-2,
    )
