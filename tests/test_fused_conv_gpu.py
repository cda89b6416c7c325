"""Numerics of the hand-written conv1 weight-gradient MFMA kernel vs
a plain fp32 PyTorch reference of the same op."""
import pytest
import torch as t
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

if not t.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)

DEV = "cuda:0"


class TestConv1Wrw:
    @pytest.mark.parametrize("batch", [1, 3, 64])
    def test_vs_fp32_reference(self, batch):
        from machin_amd.ops import _require_ext

        ext = _require_ext()
        t.manual_seed(0)
        frames = t.randint(0, 256, (batch, 84, 84, 4), dtype=t.uint8,
                           device=DEV)
        gy = (t.randn(batch, 20, 20, 32, device=DEV) * 0.1).to(t.bfloat16)

        grad_w, grad_b = ext.conv1_wrw(
            gy.reshape(-1, 32).contiguous(), frames.contiguous(),
            1.0 / 255.0,
        )
        assert grad_w.shape == (32, 4, 8, 8)
        ref_b = gy.float().sum(dim=(0, 1, 2))
        assert t.allclose(grad_b, ref_b, rtol=1e-2, atol=1e-2)

        # fp32 reference: same math, plain torch autograd
        x = (frames.permute(0, 3, 1, 2).float() / 255.0).requires_grad_(
            False
        )
        w = t.zeros(32, 4, 8, 8, device=DEV, requires_grad=True)
        y = F.conv2d(x, w, stride=4)
        # dL/dW with dL/dy = gy (NCHW)
        y.backward(gy.permute(0, 3, 1, 2).float())
        ref = w.grad
        # bf16 inputs: relative tolerance dominated by bf16 rounding
        denom = ref.abs().max().clamp_min(1e-3)
        rel_err = (grad_w - ref).abs().max() / denom
        assert rel_err.item() < 2e-2, f"rel err {rel_err.item()}"

    def test_fused_module_end_to_end(self):
        from machin_amd.ops.fused_conv import FusedAtariConv1

        t.manual_seed(1)
        B = 32
        frames = t.randint(
            0, 256, (B, 4, 84, 84), dtype=t.uint8, device=DEV
        ).to(memory_format=t.channels_last)
        stem = FusedAtariConv1().to(DEV)
        y = stem(frames)
        assert y.dtype == t.bfloat16 and y.shape == (B, 32, 20, 20)
        loss = (y.float() ** 2).sum()
        loss.backward()
        assert stem.weight.grad is not None
        assert stem.bias.grad is not None

        # compare against the plain path
        x = frames.float() / 255.0
        w = stem.weight.detach().clone().requires_grad_(True)
        b = stem.bias.detach().clone().requires_grad_(True)
        y2 = F.conv2d(
            x, w.to(t.bfloat16).float(), b.to(t.bfloat16).float(),
            stride=4,
        )
        (y2 ** 2).sum().backward()
        denom = w.grad.abs().max().clamp_min(1e-3)
        rel = (stem.weight.grad - w.grad).abs().max() / denom
        assert rel.item() < 5e-2, f"weight grad rel err {rel.item()}"
        relb = (
            (stem.bias.grad - b.grad).abs().max()
            / b.grad.abs().max().clamp_min(1e-3)
        )
        assert relb.item() < 5e-2


class TestMfmaProbe:
    def test_fragment_layout(self):
        """Asymmetric A and B (guide §3: symmetric inputs mask
        row/col swaps) through one MFMA tile vs torch matmul."""
        from machin_amd.ops import _require_ext

        ext = _require_ext()
        t.manual_seed(3)
        A = t.randn(16, 32, device=DEV).to(t.bfloat16)
        B = t.randn(32, 16, device=DEV).to(t.bfloat16)
        D = ext.mfma_probe(A.contiguous(), B.contiguous())
        ref = A.float() @ B.float()
        assert t.allclose(D, ref, rtol=2e-2, atol=1e-2), (
            (D - ref).abs().max().item()
        )


class TestConv1Fwd:
    @pytest.mark.parametrize("batch", [1, 3, 64])
    def test_vs_fp32_reference(self, batch):
        from machin_amd.ops import _require_ext

        ext = _require_ext()
        t.manual_seed(5)
        frames = t.randint(0, 256, (batch, 84, 84, 4), dtype=t.uint8,
                           device=DEV)
        w = t.randn(32, 4, 8, 8, device=DEV) * 0.05
        b = t.randn(32, device=DEV) * 0.1
        w_rs = w.permute(2, 3, 1, 0).reshape(256, 32).to(
            t.bfloat16
        ).contiguous()
        y_rows = ext.conv1_fwd(frames.contiguous(), w_rs,
                               b.contiguous(), 1.0 / 255.0)
        y = y_rows.view(batch, 20, 20, 32).permute(0, 3, 1, 2).float()
        x = frames.permute(0, 3, 1, 2).float() / 255.0
        ref = F.conv2d(x, w, b, stride=4)
        denom = ref.abs().max().clamp_min(1e-3)
        rel = (y - ref).abs().max() / denom
        assert rel.item() < 2e-2, f"rel err {rel.item()}"
