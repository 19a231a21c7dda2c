"""End-to-end training on MI355X: the full native stack (XNOR conv fwd,
dense MFMA bwd, fused BN/PReLU/kurtosis/optimizer) must actually learn.
Overfitting a small fixed batch is the convergence oracle."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_overfit_small_set_loss_drops():
    from bdbnn_amd.models import imagenet as im
    from bdbnn_amd.ops.optim import FusedAdam
    torch.manual_seed(0)
    model = im.resnet18(False, num_classes=10).cuda().to(
        memory_format=torch.channels_last)
    x = torch.randn(32, 3, 64, 64, device="cuda").contiguous(
        memory_format=torch.channels_last)
    y = torch.randint(0, 10, (32,), device="cuda")
    opt = FusedAdam(model.parameters(), lr=3e-3)
    losses = []
    for i in range(60):
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = model(x)
            loss = torch.nn.functional.cross_entropy(out, y)
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    first = sum(losses[:5]) / 5
    last = sum(losses[-5:]) / 5
    assert last < first * 0.5, (first, last)
    assert all(l == l for l in losses)  # no NaNs


def test_trainer_step_gpu_with_kurtosis_and_ts():
    from train import build_parser, build_model, build_teacher
    from bdbnn_amd.engine import Trainer
    args = build_parser().parse_args([
        "./", "--dataset", "imagenet", "-a", "resnet18", "-b", "16",
        "--epochs", "1", "--w-kurtosis", "--weight-name", "all",
        "--diffkurt", "--imagenet_setting_step_2_ts",
        "-a_teacher", "resnet18", "--amp"])
    model = build_model(args, "resnet18", True)
    teacher = build_teacher(args)
    trainer = Trainer(model, args, teacher=teacher,
                      device=torch.device("cuda"))
    x = torch.randn(16, 3, 64, 64, device="cuda").contiguous(
        memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (16,), device="cuda")
    with torch.autocast("cuda", dtype=torch.bfloat16):
        total, ce, kurt, out = trainer._step_losses_ts(x, y, 0)
    total.backward()
    trainer.model.finish_gradient_sync()
    trainer.optimizer.step()
    assert torch.isfinite(total)
    assert kurt is not None and torch.isfinite(kurt)


def test_ede_epoch_injection_gpu():
    from bdbnn_amd.models import imagenet as im
    from bdbnn_amd.engine.trainer import ede_inject
    model = im.resnet18(False).cuda().to(memory_format=torch.channels_last)
    ede_inject(model, 10, 100)
    x = torch.randn(2, 3, 64, 64, device="cuda").contiguous(
        memory_format=torch.channels_last)
    out = model(x)
    out.sum().backward()
    assert torch.isfinite(out).all()


def test_learns_separable_classes():
    """Generalization (not just memorization): gaussian class clusters in
    image space; a binarized ResNet-20 must beat chance on HELD-OUT
    samples after a short training run on the native stack."""
    from bdbnn_amd.models import cifar10 as cm
    from bdbnn_amd.ops.optim import FusedSGD
    torch.manual_seed(1)
    n_classes = 10
    means = torch.randn(n_classes, 3, 32, 32, device="cuda") * 1.5

    def sample(n):
        y = torch.randint(0, n_classes, (n,), device="cuda")
        x = means[y] + torch.randn(n, 3, 32, 32, device="cuda") * 0.5
        return x.contiguous(memory_format=torch.channels_last), y

    model = cm.resnet20().cuda().to(memory_format=torch.channels_last)
    opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9)
    model.train()
    for i in range(150):
        x, y = sample(128)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = torch.nn.functional.cross_entropy(model(x), y)
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
    model.eval()
    with torch.no_grad():
        x, y = sample(512)
        acc = (model(x).argmax(1) == y).float().mean().item()
    assert acc > 0.35, acc  # chance = 0.1
