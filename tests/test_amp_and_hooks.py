import math

import pytest
import torch

from tests.utils_dist import run_distributed

pytestmark = pytest.mark.timeout(300)


def test_grad_scaler_skips_on_overflow():
    from pdnlp_amd.amp import GradScaler
    p = torch.nn.Parameter(torch.ones(4))
    opt = torch.optim.SGD([p], lr=1.0)
    scaler = GradScaler(init_scale=8.0)
    loss = (p * torch.tensor([1.0, 1.0, 1.0, 1.0])).sum()
    scaler.scale(loss).backward()
    p.grad[0] = float("inf")
    before = p.detach().clone()
    scaler.step(opt)
    scaler.update()
    torch.testing.assert_close(p.detach(), before)  # step skipped
    assert scaler.get_scale() == 4.0  # backoff 0.5


def test_grad_scaler_unscales():
    from pdnlp_amd.amp import GradScaler
    p = torch.nn.Parameter(torch.zeros(3))
    opt = torch.optim.SGD([p], lr=1.0)
    scaler = GradScaler(init_scale=16.0)
    loss = p.sum()
    scaler.scale(loss).backward()
    assert torch.allclose(p.grad, torch.full((3,), 16.0))
    scaler.step(opt)
    scaler.update()
    torch.testing.assert_close(p.detach(), torch.full((3,), -1.0))


def test_fused_adamw_matches_torch_adamw():
    from pdnlp_amd.ops.adamw import FusedAdamW
    torch.manual_seed(0)
    w1 = torch.nn.Parameter(torch.randn(16, 16))
    w2 = torch.nn.Parameter(torch.randn(16, 16))
    with torch.no_grad():
        w2.copy_(w1)
    opt1 = FusedAdamW([w1], lr=1e-3, weight_decay=0.01)
    opt2 = torch.optim.AdamW([w2], lr=1e-3, weight_decay=0.01)
    for i in range(5):
        g = torch.randn(16, 16)
        w1.grad = g.clone()
        w2.grad = g.clone()
        opt1.step()
        opt2.step()
    torch.testing.assert_close(w1, w2, rtol=1e-5, atol=1e-7)


def test_fused_adamw_bf16_master_weights():
    from pdnlp_amd.ops.adamw import FusedAdamW
    torch.manual_seed(0)
    w = torch.nn.Parameter(torch.randn(64).bfloat16())
    opt = FusedAdamW([w], lr=1e-2)
    for _ in range(3):
        w.grad = torch.randn(64).bfloat16()
        opt.step()
    st = opt.state[w]
    assert st["master"] is not None and st["master"].dtype == torch.float32
    torch.testing.assert_close(w.detach(), st["master"].bfloat16())


def _hooks_optimizer(rank, world):
    from pdnlp_amd.ops.adamw import build_optimizer
    from pdnlp_amd.parallel import (DistributedOptimizer,
                                    broadcast_parameters)
    from pdnlp_amd.config import BertConfig
    from pdnlp_amd.models import BertForSequenceClassification
    from pdnlp_amd.utils import set_seed
    set_seed(123 + rank)  # deliberately different init; broadcast must fix it
    cfg = BertConfig.tiny()
    model = BertForSequenceClassification(cfg)
    broadcast_parameters(model)
    sd = [model.classifier.weight.detach().clone()]
    gather = [torch.zeros_like(sd[0]) for _ in range(world)]
    torch.distributed.all_gather(gather, sd[0])
    torch.testing.assert_close(gather[0], gather[1])  # broadcast worked

    opt = DistributedOptimizer(build_optimizer(model, lr=1e-3))
    g = torch.Generator().manual_seed(50 + rank)
    ids = torch.randint(0, cfg.vocab_size, (4, 16), generator=g)
    mask = torch.ones(4, 16, dtype=torch.long)
    labels = torch.randint(0, cfg.num_labels, (4,), generator=g)
    out = model(ids, mask, labels=labels)
    opt.zero_grad(set_to_none=False)
    out.loss.backward()
    opt.step()
    # params identical across ranks after hooked allreduce + step
    w = model.classifier.weight.detach().clone()
    gather = [torch.zeros_like(w) for _ in range(world)]
    torch.distributed.all_gather(gather, w)
    torch.testing.assert_close(gather[0], gather[1], rtol=1e-5, atol=1e-7)


def test_hooks_distributed_optimizer():
    run_distributed(_hooks_optimizer, world=2)
