"""Oracle equality: differential attention / DiffTransformer vs the reference.

The reference (replay/nn/attention.py:7, replay/nn/sequential/sasrec/
diff_transformer.py:10,67) runs in place as the numerical oracle; our modules
must load its state dicts verbatim and produce identical outputs.
"""

import pytest
import torch

pytestmark = pytest.mark.core


def import_reference_module(name: str):
    import importlib
    import os
    import sys

    sys.path.insert(0, os.path.dirname(__file__))
    from _reference_harness import load_reference

    load_reference()
    return importlib.import_module(name)


def _rand_mask(B, H, L):
    torch.manual_seed(3)
    causal = torch.tril(torch.ones(L, L, dtype=torch.bool))
    mask = torch.zeros(B * H, L, L)
    mask = mask.masked_fill(~causal[None], float("-inf"))
    return mask


def test_diff_attention_state_dict_and_numerics():
    ref_mod = import_reference_module("replay.nn.attention")
    from replay_amd.nn.attention import MultiHeadDifferentialAttention

    B, L, E, H = 3, 7, 16, 2
    torch.manual_seed(0)
    ref = ref_mod.MultiHeadDifferentialAttention(E, H, lambda_init=0.5, vdim=2 * E)
    torch.manual_seed(1)
    ours = MultiHeadDifferentialAttention(E, H, lambda_init=0.5, vdim=2 * E)

    # state dicts are interchangeable (same keys, same shapes)
    ours.load_state_dict(ref.state_dict())

    torch.manual_seed(2)
    x = torch.randn(B, L, E)
    mask = _rand_mask(B, H, L)
    out_ref = ref(x, x, x, mask)
    out_ours = ours(x, x, x, mask)
    torch.testing.assert_close(out_ours, out_ref, rtol=1e-5, atol=1e-6)


def test_diff_transformer_layer_oracle():
    ref_mod = import_reference_module("replay.nn.sequential.sasrec.diff_transformer")
    from replay_amd.nn.sequential.sasrec.diff_transformer import DiffTransformerLayer

    B, L, E, H = 2, 5, 16, 2
    torch.manual_seed(0)
    ref = ref_mod.DiffTransformerLayer(embedding_dim=E, num_heads=H, num_blocks=2)
    ours = DiffTransformerLayer(E, H, num_blocks=2)
    ours.load_state_dict(ref.state_dict())

    torch.manual_seed(4)
    x = torch.randn(B, L, E)
    mask = _rand_mask(B, H, L)
    out_ref = ref(feature_tensors={}, input_embeddings=x, padding_mask=None, attention_mask=mask)
    out_ours = ours(x, attn_mask=mask)
    torch.testing.assert_close(out_ours, out_ref, rtol=1e-5, atol=1e-6)


def test_swiglu_state_dict_matches_reference():
    ref_mod = import_reference_module("replay.nn.ffn")
    from replay_amd.nn.ffn import SwiGLU, SwiGLUEncoder

    E = 12
    torch.manual_seed(0)
    ref = ref_mod.SwiGLU(E, 2 * E)
    ours = SwiGLU(E, 2 * E)
    ours.load_state_dict(ref.state_dict())
    x = torch.randn(4, 6, E)
    torch.testing.assert_close(ours(x), ref(x), rtol=1e-6, atol=1e-7)

    torch.manual_seed(0)
    ref_enc = ref_mod.SwiGLUEncoder(E, 2 * E)
    ours_enc = SwiGLUEncoder(E, 2 * E)
    ours_enc.load_state_dict(ref_enc.state_dict())
    torch.testing.assert_close(
        ours_enc(x), ref_enc(feature_tensors={}, input_embeddings=x), rtol=1e-6, atol=1e-7
    )
