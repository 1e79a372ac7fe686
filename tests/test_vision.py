"""ViT vision encoder (K12) + ColPali-style late interaction."""
import torch

from modal_examples_amd.models.vision import ViTConfig, VisionEncoder, maxsim


def test_embed_shapes_and_norm():
    enc = VisionEncoder(ViTConfig.small_test()).eval()
    imgs = torch.randn(3, 3, 64, 64)
    e = enc.embed(imgs)
    assert e.shape == (3, 16, 32)  # 64/16=4 -> 16 patches, embed 32
    n = e.norm(dim=-1)
    assert torch.allclose(n, torch.ones_like(n), atol=1e-4)


def test_maxsim_retrieves_perturbed_page():
    torch.manual_seed(0)
    enc = VisionEncoder(ViTConfig.small_test()).eval()
    pages = torch.randn(5, 3, 64, 64)
    embs = enc.embed(pages)
    q = enc.embed((pages[3] + 0.1 * torch.randn_like(pages[3]))[None])[0]
    scores = maxsim(q, embs)
    assert int(scores.argmax()) == 3
