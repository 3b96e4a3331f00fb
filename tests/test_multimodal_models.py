"""ViT / CoCa / generic attention+MLP tests (reference model:
tests for models/vision_transformer and models/coca)."""

import torch

from modalities_amd.batch import InferenceResultBatch
from modalities_amd.loss_functions import NCELoss
from modalities_amd.models.coca import CoCa
from modalities_amd.models.vision_transformer import VisionTransformer
from modalities_amd.nn.attention import (AttentionType, MultiHeadAttention)


def test_multihead_attention_modes():
    torch.manual_seed(0)
    x = torch.randn(2, 8, 32)
    ctx = torch.randn(2, 12, 32)
    for at in AttentionType:
        attn = MultiHeadAttention(32, 4, attention_type=at)
        y = attn(x, context=ctx)
        assert y.shape == x.shape
    # causal: output at position t must not depend on inputs > t
    attn = MultiHeadAttention(32, 4, attention_type=AttentionType.CAUSAL_SELF_ATTENTION)
    attn.eval()
    y1 = attn(x)
    x2 = x.clone()
    x2[:, -1] += 100.0
    y2 = attn(x2)
    torch.testing.assert_close(y1[:, :-1], y2[:, :-1])


def test_vision_transformer_shapes():
    torch.manual_seed(0)
    vit = VisionTransformer(img_size=64, n_layer=2, n_head=4, n_embd=64,
                            ffn_hidden=128, n_classes=10, patch_size=16,
                            patch_stride=16)
    imgs = torch.randn(2, 3, 64, 64)
    out = vit({"images": imgs})
    assert out["cls_token"].shape == (2, 10)
    # mean-pool variant without classes
    vit2 = VisionTransformer(img_size=64, n_layer=1, n_head=4, n_embd=64,
                             ffn_hidden=128, n_classes=None,
                             add_cls_token=False)
    out2 = vit2({"images": imgs})
    assert out2["cls_token"].shape == (2, 64)


def test_coca_forward_backward_and_nce():
    torch.manual_seed(0)
    model = CoCa(vocab_size=256, text_block_size=16, n_layer_text=1,
                 n_layer_multimodal_text=1, n_head=4, n_embd=64,
                 ffn_hidden=128, n_vision_queries=8,
                 vision_encoder_config={"img_size": 32, "n_layer": 1,
                                        "patch_size": 16, "patch_stride": 16})
    images = torch.randn(2, 3, 32, 32)
    ids = torch.randint(0, 256, (2, 16))
    out = model({"images": images, "input_ids": ids})
    assert out["logits"].shape == (2, 16, 256)
    assert out["vision_cls"].shape == (2, 64)
    assert out["text_cls"].shape == (2, 64)

    batch = InferenceResultBatch(targets={}, predictions=out)
    nce = NCELoss("vision_cls", "text_cls")
    cap = torch.nn.functional.cross_entropy(
        out["logits"].reshape(-1, 256), ids.reshape(-1))
    loss = nce(batch) + cap
    loss.backward()
    assert model.vision_encoder.blocks[0].attention.wq.weight.grad is not None
    assert model.text_decoder.blocks[0].attn.wq.weight.grad is not None
