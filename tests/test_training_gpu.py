"""GPU training tier: LoRA gradients flow end-to-end on MI355X (the
training-mode dispatch uses differentiable math; fused AdamW updates)."""
import pytest
import torch

pytestmark = pytest.mark.gpu
requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")


@requires_gpu
def test_lora_train_step_on_gpu():
    from modal_examples_amd.models.sdxl.unet import UNetConfig
    from modal_examples_amd.train.dreambooth import LoRATrainer, TrainConfig

    t = LoRATrainer(
        UNetConfig.small(),
        TrainConfig(rank=4, batch_size=2, resolution=128, max_steps=3),
        device="cuda", dtype=torch.bfloat16)
    l1 = t.train_step()
    assert l1 > 0
    moved = sum(float(p.abs().sum()) for p in t.params)
    assert moved > 0, "LoRA params did not update"
    l2 = t.train_step()
    assert t.step_count == 2 and l2 > 0


@requires_gpu
def test_gpt_train_step_on_gpu():
    from modal_examples_amd.models.gpt.model import GPT, GPTConfig
    from modal_examples_amd.train.lora import FusedAdamW

    torch.manual_seed(0)
    cfg = GPTConfig(n_layer=2, n_embd=128, n_head=2, block_size=64)
    model = GPT(cfg).to("cuda", torch.bfloat16)
    opt = FusedAdamW(list(model.parameters()), lr=1e-3)
    x = torch.randint(0, cfg.vocab_size, (4, 64), device="cuda")
    losses = []
    for _ in range(8):
        _, loss = model(x, torch.roll(x, -1, 1))
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0], f"no learning: {losses[0]:.3f}→{losses[-1]:.3f}"
