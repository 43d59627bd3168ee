"""Model zoo smoke tests (workloads for the runtime's GPU paths)."""
import torch

from lzy_amd.models import TransformerLM, resnet50


def test_resnet50_forward_backward():
    m = resnet50(num_classes=10)
    x = torch.randn(2, 3, 64, 64)
    y = m(x)
    assert y.shape == (2, 10)
    y.square().mean().backward()
    n_params = sum(p.numel() for p in m.parameters())
    assert 20_000_000 < n_params < 30_000_000  # ResNet-50 scale


def test_transformer_lm_loss_backward():
    m = TransformerLM(vocab=128, d_model=64, n_layers=2, n_heads=4, max_seq=32)
    idx = torch.randint(0, 128, (2, 16))
    logits = m(idx)
    assert logits.shape == (2, 16, 128)
    loss = m.loss(idx)
    assert loss.isfinite()
    loss.backward()
    assert m.tok.weight.grad is not None


def test_transformer_in_workflow(tmp_path, monkeypatch):
    monkeypatch.setenv("LZY_AMD_STORAGE", str(tmp_path / "s"))
    from lzy_amd import Lzy, op
    from lzy_amd.runtime.local import LocalRuntime

    @op
    def train_tiny(steps: int) -> float:
        m = TransformerLM(vocab=64, d_model=32, n_layers=1, n_heads=2, max_seq=16)
        opt = torch.optim.AdamW(m.parameters(), lr=1e-3)
        loss = None
        for _ in range(steps):
            idx = torch.randint(0, 64, (2, 12))
            opt.zero_grad(set_to_none=True)
            loss = m.loss(idx)
            loss.backward()
            opt.step()
        return float(loss.item())

    with Lzy(runtime=LocalRuntime()).workflow("lm", interactive=False):
        l = train_tiny(3)
        assert float(l) > 0
