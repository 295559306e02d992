import pytest
import torch

from murmura_amd.models import (
    CelebAModel,
    EvidentialHARClassifier,
    EvidentialLoss,
    EvidentialPAMAP2Classifier,
    EvidentialPPGDaLiAClassifier,
    FEMNISTModel,
    ResNet18,
    SimpleMLP,
    compute_uncertainty,
    count_params,
    get_evidential_loss,
    get_model_variant,
)
from murmura_amd.models.zoo import WideMLP


@pytest.mark.parametrize("variant,lo,hi", [
    ("tiny", 0.1e6, 0.35e6),
    ("small", 0.6e6, 1.0e6),
    ("baseline", 6.0e6, 7.0e6),
    ("large", 12.5e6, 14.0e6),
    ("xlarge", 25.0e6, 27.5e6),
])
def test_femnist_size_ladder(variant, lo, hi):
    n = count_params(get_model_variant(variant))
    assert lo < n < hi, (variant, n)


def test_femnist_forward():
    m = FEMNISTModel(hidden=64)
    out = m(torch.randn(4, 1, 28, 28))
    assert out.shape == (4, 62)


def test_celeba_forward():
    out = CelebAModel()(torch.randn(2, 3, 84, 84))
    assert out.shape == (2, 2)


def test_widemlp_is_100m():
    n = count_params(WideMLP())
    assert 95e6 < n < 110e6


@pytest.mark.parametrize("cls,shape,classes", [
    (EvidentialHARClassifier, (8, 561), 6),
    (EvidentialPAMAP2Classifier, (8, 4000), 12),
    (EvidentialPPGDaLiAClassifier, (8, 192), 7),
])
def test_evidential_classifiers(cls, shape, classes):
    m = cls()
    m.eval()
    out = m(torch.randn(*shape))
    assert out.shape == (shape[0], classes)


def test_compute_uncertainty_bounds():
    u = compute_uncertainty(torch.randn(16, 6))
    assert torch.all(u["vacuity"] > 0) and torch.all(u["vacuity"] <= 1.0)
    assert torch.all(u["strength"] >= 6.0)
    assert torch.allclose(u["probs"].sum(1), torch.ones(16), atol=1e-5)


def test_evidential_loss_anneals():
    loss_fn = EvidentialLoss(num_classes=6, annealing_rounds=10, max_kl_weight=0.1)
    logits = torch.randn(32, 6)
    y = torch.randint(0, 6, (32,))
    l0 = loss_fn(logits, y, round_num=0)
    l10 = loss_fn(logits, y, round_num=10)
    assert l10 >= l0  # KL term ramps in
    assert torch.isfinite(l0) and torch.isfinite(l10)


def test_evidential_loss_trains():
    torch.manual_seed(0)
    m = SimpleMLP(10, 32, 4)
    loss_fn = get_evidential_loss(4, total_rounds=20)
    x = torch.randn(256, 10)
    y = (x[:, 0] > 0).long() + 2 * (x[:, 1] > 0).long()
    opt = torch.optim.SGD(m.parameters(), lr=0.5)
    first = None
    for step in range(60):
        opt.zero_grad()
        loss = loss_fn(m(x), y, round_num=0)
        if first is None:
            first = loss.item()
        loss.backward()
        opt.step()
    assert loss.item() < first * 0.8


def test_correct_prediction_lowers_evidential_loss():
    loss_fn = EvidentialLoss(num_classes=3)
    y = torch.tensor([0])
    right = torch.tensor([[5.0, -5.0, -5.0]])
    wrong = torch.tensor([[-5.0, 5.0, -5.0]])
    assert loss_fn(right, y, 0) < loss_fn(wrong, y, 0)


def test_basic_block_residual_fusion_cpu_semantics():
    """BasicBlock's fused tail relu(bn2(conv2)+shortcut) must equal the plain
    composition on the CPU fallback path (the GPU kernel is compared to torch
    in tests/test_gpu_kernels.py)."""
    import torch
    from torch import nn

    from murmura_amd.models.zoo import BasicBlock

    torch.manual_seed(0)
    blk = BasicBlock(8, 8)
    blk.train()
    x = torch.randn(4, 8, 6, 6)
    out = blk(x)

    # reference composition using the same parameters
    ref_bn2 = nn.BatchNorm2d(8)
    ref_bn2.load_state_dict(
        {k.split("bn2.")[1]: v for k, v in blk.state_dict().items()
         if k.startswith("bn2.")}
    )
    torch.manual_seed(0)
    blk2 = BasicBlock(8, 8)
    blk2.load_state_dict(blk.state_dict())
    h = blk2.bn1(blk2.conv1(x))
    expected = torch.relu(ref_bn2(blk2.conv2(h)) + x)
    assert torch.allclose(out, expected, atol=1e-5)
