"""fp8 (e4m3/e5m2) linear-path numerics on MI355X: _scaled_mm micro-GEMM
vs the bf16 reference, and one full fp8-mode training step (VERDICT r1
item 9: exercise the measured fp8 shapes end-to-end)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _rel(a, b):
    return ((a.float() - b.float()).norm() / b.float().norm().clamp(min=1e-9)).item()


def test_fp8_linear_micro_step():
    from simumax_amd.kernels.fp8 import Fp8Linear, fp8_available

    if not fp8_available():
        pytest.skip("no _scaled_mm fp8 support")
    torch.manual_seed(0)
    lin = Fp8Linear(4096, 2048, device=DEV)
    x = torch.randn(8, 512, 4096, device=DEV, dtype=torch.bfloat16,
                    requires_grad=True)
    y = lin(x)
    # bf16 reference with identical weights
    xr = x.detach().clone().requires_grad_(True)
    yr = xr @ lin.weight.detach().t()
    assert _rel(y, yr) < 0.05, f"fwd rel {_rel(y, yr)}"
    dy = torch.randn_like(y)
    y.backward(dy)
    yr.backward(dy)
    assert _rel(x.grad, xr.grad) < 0.12, f"dgrad rel {_rel(x.grad, xr.grad)}"
    wg_ref = dy.reshape(-1, 2048).t().float() @ xr.reshape(-1, 4096).float()
    assert _rel(lin.weight.grad, wg_ref) < 0.12, \
        f"wgrad rel {_rel(lin.weight.grad, wg_ref)}"


def test_fp8_trainer_step():
    from simumax_amd.core.config import ModelConfig
    from simumax_amd.kernels.fp8 import fp8_available
    from simumax_amd.train.trainer import (TrainConfig, build_trainer,
                                           make_synthetic_batch, train_step)

    if not fp8_available():
        pytest.skip("no _scaled_mm fp8 support")
    cfg = ModelConfig(hidden_size=512, head_num=4, kv_head_num=2,
                      head_size=128, intermediate_size=1024, layer_num=2,
                      vocab_size=2048, use_swiglu=True)
    tc = TrainConfig(seq_len=512, micro_batch_size=2, micro_batch_num=2,
                     fp8=True)
    model, opt, red = build_trainer(cfg, tc, DEV)
    toks, labels = make_synthetic_batch(cfg.vocab_size, 2, 2, 512, DEV)
    l0 = train_step(model, opt, red, toks, labels, 2)
    l1 = train_step(model, opt, red, toks, labels, 2)
    assert l0 == l0 and l1 == l1, "fp8 loss NaN"
    assert l1 < l0 + 0.5  # training is not diverging on repeated batch
