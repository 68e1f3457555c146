"""GPU (MI355X) numerics tests: every HIP kernel against its torch fp32
oracle (methodology: ESR:models/DCNv2/testcuda.py — CPU path is the oracle).

All tests here are @pytest.mark.gpu and require the in-tree _esr_native
extension; they FAIL (not skip) if the extension is missing on a GPU box.
"""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from esr_amd.ops.native import get_ext
    e = get_ext()
    assert e is not None, "native extension not built on a GPU box"
    return e


def _dcn_problem(B=2, C=16, H=14, W=18, Cout=12, dg=4, seed=0, dev="cuda"):
    g = torch.Generator().manual_seed(seed)
    input = torch.randn(B, C, H, W, generator=g).to(dev)
    offset = (torch.randn(B, dg * 18, H, W, generator=g) * 2).to(dev)
    mask = torch.rand(B, dg * 9, H, W, generator=g).to(dev)
    weight = (torch.randn(Cout, C, 3, 3, generator=g) * 0.2).to(dev)
    bias = torch.randn(Cout, generator=g).to(dev)
    return input, offset, mask, weight, bias


class TestDeformConv:
    def test_forward_matches_oracle(self, ext):
        from esr_amd.ops.dcn import _deform_conv2d_torch
        input, offset, mask, weight, bias = _dcn_problem()
        out = ext.deform_conv2d_forward(input, offset, mask, weight, bias,
                                        1, 1, 1, 1, 1, 1, 4)
        ref = _deform_conv2d_torch(input.cpu(), offset.cpu(), mask.cpu(),
                                   weight.cpu(), bias.cpu(),
                                   (1, 1), (1, 1), (1, 1), 4)
        assert torch.allclose(out.cpu(), ref, atol=1e-4), \
            (out.cpu() - ref).abs().max().item()

    def test_forward_zero_offset_is_conv(self, ext):
        import torch.nn.functional as F
        input, _, _, weight, bias = _dcn_problem()
        offset = torch.zeros(2, 4 * 18, 14, 18, device="cuda")
        mask = torch.ones(2, 4 * 9, 14, 18, device="cuda")
        out = ext.deform_conv2d_forward(input, offset, mask, weight, bias,
                                        1, 1, 1, 1, 1, 1, 4)
        ref = F.conv2d(input, weight, bias, 1, 1)
        assert torch.allclose(out, ref, atol=1e-4)

    def test_backward_matches_oracle(self, ext):
        from esr_amd.ops.dcn import _deform_conv2d_torch
        input, offset, mask, weight, bias = _dcn_problem(seed=3)
        args_cpu = [t.cpu().detach().requires_grad_(True)
                    for t in (input, offset, mask, weight, bias)]
        ref_out = _deform_conv2d_torch(*args_cpu, (1, 1), (1, 1), (1, 1), 4)
        gout = torch.randn_like(ref_out)
        ref_out.backward(gout)
        grads = ext.deform_conv2d_backward(input, offset, mask, weight,
                                           gout.cuda().contiguous(),
                                           1, 1, 1, 1, 1, 1, 4)
        names = ["input", "offset", "mask", "weight", "bias"]
        for g_hip, t_cpu, name in zip(grads, args_cpu, names):
            err = (g_hip.cpu() - t_cpu.grad).abs().max().item()
            scale = t_cpu.grad.abs().max().item() + 1e-6
            assert err / scale < 1e-3, f"grad_{name} rel err {err/scale}"

    def test_autograd_path_end_to_end(self, ext):
        from esr_amd.ops.dcn import modulated_deform_conv2d
        input, offset, mask, weight, bias = _dcn_problem(seed=5)
        for t in (input, offset, mask, weight, bias):
            t.requires_grad_(True)
        out = modulated_deform_conv2d(input, offset, mask, weight, bias,
                                      stride=1, padding=1, dilation=1,
                                      deformable_groups=4)
        out.square().mean().backward()
        for t in (input, offset, mask, weight, bias):
            assert t.grad is not None and torch.isfinite(t.grad).all()

    def test_stride2(self, ext):
        from esr_amd.ops.dcn import _deform_conv2d_torch
        g = torch.Generator().manual_seed(9)
        input = torch.randn(1, 8, 16, 16, generator=g).cuda()
        offset = (torch.randn(1, 2 * 18, 8, 8, generator=g)).cuda()
        mask = torch.rand(1, 2 * 9, 8, 8, generator=g).cuda()
        weight = torch.randn(4, 8, 3, 3, generator=g).cuda() * 0.2
        out = ext.deform_conv2d_forward(input, offset, mask, weight, None,
                                        2, 2, 1, 1, 1, 1, 2)
        ref = _deform_conv2d_torch(input.cpu(), offset.cpu(), mask.cpu(),
                                   weight.cpu(), None, (2, 2), (1, 1), (1, 1), 2)
        assert out.shape == (1, 4, 8, 8)
        assert torch.allclose(out.cpu(), ref, atol=1e-4)


class TestGruGates:
    @pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
    def test_ur_forward(self, ext, dtype):
        g = torch.Generator().manual_seed(0)
        ur = torch.randn(2, 8, 12, 12, generator=g).cuda().to(dtype)
        h = torch.randn(2, 4, 12, 12, generator=g).cuda().to(dtype)
        u, r, hr = ext.gru_gates_ur_forward(ur.contiguous(), h.contiguous())
        uf = torch.sigmoid(ur[:, :4].float())
        rf = torch.sigmoid(ur[:, 4:].float())
        tol = 1e-5 if dtype == torch.float32 else 2e-2
        assert torch.allclose(u.float(), uf, atol=tol)
        assert torch.allclose(r.float(), rf, atol=tol)
        assert torch.allclose(hr.float(), (h.float() * rf), atol=tol)

    def test_gates_backward_matches_autograd(self, ext):
        torch.manual_seed(1)
        ur = torch.randn(2, 8, 10, 10, device="cuda", requires_grad=True)
        h = torch.randn(2, 4, 10, 10, device="cuda", requires_grad=True)
        o_pre = torch.randn(2, 4, 10, 10, device="cuda", requires_grad=True)

        # torch autograd reference
        u_ref = torch.sigmoid(ur[:, :4])
        r_ref = torch.sigmoid(ur[:, 4:])
        h_new_ref = h * (1 - u_ref) + torch.tanh(o_pre) * u_ref
        # also route a loss through hr to exercise that path
        hr_ref = h * r_ref
        loss_ref = h_new_ref.square().mean() + hr_ref.mean()
        loss_ref.backward()

        from esr_amd.ops.convgru import gru_gates_ur, gru_gates_out
        ur2 = ur.detach().clone().requires_grad_(True)
        h2 = h.detach().clone().requires_grad_(True)
        o2 = o_pre.detach().clone().requires_grad_(True)
        u, r, hr = gru_gates_ur(ur2, h2)
        h_new = gru_gates_out(o2, u, h2)
        loss = h_new.square().mean() + hr.mean()
        loss.backward()

        for a, b, name in [(ur.grad, ur2.grad, "ur"), (h.grad, h2.grad, "h"),
                           (o_pre.grad, o2.grad, "o_pre")]:
            assert torch.allclose(a, b, atol=1e-5), \
                f"{name}: {(a-b).abs().max().item()}"

    def test_cell_gpu_matches_cpu(self, ext):
        from esr_amd.ops.convgru import ConvGRUCell
        torch.manual_seed(2)
        cell = ConvGRUCell(4, 4, 3)
        x = torch.randn(2, 4, 16, 16)
        h1 = cell(x, None)
        h2 = cell(x, h1)
        cell_g = cell.cuda()
        h1g = cell_g(x.cuda(), None)
        h2g = cell_g(x.cuda(), h1g)
        assert torch.allclose(h2g.cpu(), h2, atol=1e-4)


class TestSplat:
    def test_splat_count_matches_torch(self, ext):
        from esr_amd.ops import events_to_channels
        g = torch.Generator().manual_seed(0)
        B, N, H, W = 3, 4096, 32, 48
        ev = torch.zeros(B, N, 4)
        ev[:, :, 0] = torch.rand(B, N, generator=g) * (W + 4) - 2
        ev[:, :, 1] = torch.rand(B, N, generator=g) * (H + 4) - 2
        ev[:, :, 2] = torch.sort(torch.rand(B, N, generator=g), dim=1).values
        ev[:, :, 3] = torch.randint(0, 2, (B, N), generator=g) * 2 - 1
        ev[:, N - 100:, 3] = 0  # padding
        ev[:, :, 0] = ev[:, :, 0].floor()
        ev[:, :, 1] = ev[:, :, 1].floor()
        out = ext.splat_count(ev.cuda().contiguous(), H, W)
        for b in range(B):
            e = ev[b]
            e = e[e[:, 3] != 0]
            ref = events_to_channels(e[:, 0], e[:, 1], e[:, 3], (H, W))
            assert torch.allclose(out[b].cpu(), ref, atol=1e-4)

    def test_splat_stack_matches_torch(self, ext):
        from esr_amd.ops import events_to_stack_no_polarity
        g = torch.Generator().manual_seed(1)
        B, N, TB, H, W = 2, 2048, 4, 16, 16
        ev = torch.zeros(B, N, 4)
        ev[:, :, 0] = (torch.rand(B, N, generator=g) * W).floor()
        ev[:, :, 1] = (torch.rand(B, N, generator=g) * H).floor()
        ev[:, :, 2] = torch.sort(torch.rand(B, N, generator=g), dim=1).values
        ev[:, :, 3] = torch.randint(0, 2, (B, N), generator=g) * 2 - 1
        out = ext.splat_stack(ev.cuda().contiguous(), TB, H, W, 0.0, 1.0)
        # bin edges differ (kernel uses fixed [0,1] range, floor binning);
        # compare totals per pixel instead of per-bin exactness
        for b in range(B):
            e = ev[b]
            ref = events_to_stack_no_polarity(e[:, 0], e[:, 1], e[:, 2],
                                              e[:, 3], TB, (H, W))
            assert torch.allclose(out[b].sum(0).cpu(), ref.sum(0), atol=1e-4)


class TestModelGPU:
    def test_model_forward_backward_gpu(self, ext):
        from esr_amd.models import build_model
        m = build_model("ESRNet", inch=2, basech=8, num_frame=3).cuda()
        x = torch.randn(2, 3, 2, 64, 64, device="cuda")
        m.reset_states()
        loss = 0
        for _ in range(2):
            loss = loss + (m(x) ** 2).mean()
        loss.backward()
        assert all(torch.isfinite(p.grad).all()
                   for p in m.parameters() if p.grad is not None)

    def test_model_bf16_autocast(self, ext):
        from esr_amd.models import build_model
        m = build_model("ESRNet", inch=2, basech=8, num_frame=3,
                        upsampler="pixelshuffle").cuda()
        x = torch.randn(2, 3, 2, 64, 64, device="cuda")
        m.reset_states()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            y = m(x)
        loss = y.float().square().mean()
        loss.backward()
        assert torch.isfinite(loss).item()

    def test_cpu_gpu_forward_parity(self, ext):
        from esr_amd.models import build_model
        torch.manual_seed(0)
        m = build_model("ESRNet", inch=2, basech=8, num_frame=3)
        x = torch.randn(1, 3, 2, 32, 32)
        m.reset_states()
        y_cpu = m(x)
        mg = m.cuda()
        mg.reset_states()
        y_gpu = mg(x.cuda())
        assert torch.allclose(y_cpu, y_gpu.cpu(), atol=5e-3), \
            (y_cpu - y_gpu.cpu()).abs().max().item()


class TestBenchPath:
    def test_bench_graph_step(self, ext):
        """End-to-end bench step (graph capture + replay) as a subprocess."""
        import json
        import subprocess
        import sys
        out = subprocess.run(
            [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
             "--batch", "2"],
            capture_output=True, text=True, timeout=600,
            cwd=str(__import__("pathlib").Path(__file__).parent.parent))
        assert out.returncode == 0, out.stderr[-2000:]
        line = [ln for ln in out.stdout.splitlines() if ln.startswith("{")][-1]
        rec = json.loads(line)
        assert rec["value"] > 0
        assert rec["config"]["hip_graphs"] == "full"

    def test_redistribute_on_gpu(self, ext):
        from esr_amd.ops import redistribute_stack, events_to_stack_no_polarity
        g = torch.Generator().manual_seed(0)
        stack = torch.randint(-3, 8, (2, 4, 8, 8), generator=g).float().cuda()
        cloud = redistribute_stack(stack, mode="linear")
        assert cloud.is_cuda
        # round trip on device
        for b in range(2):
            ev = cloud[b]
            ev = ev[ev.abs().sum(1) > 0]
            rebuilt = torch.zeros(4, 8, 8, device="cuda")
            c = ((ev[:, 2] - 1e-6) * 4).long().clamp(0, 3)
            idx = c * 64 + ev[:, 1].long() * 8 + ev[:, 0].long()
            rebuilt.view(-1).scatter_add_(0, idx, ev[:, 3])
            assert torch.allclose(rebuilt, stack[b])


class TestFusedDCN:
    """Fused im2col+MFMA forward vs the generic im2col+GEMM path."""

    @pytest.mark.parametrize("B,C,H,W,Cout,dg", [
        (2, 16, 16, 16, 16, 4),
        (4, 64, 32, 32, 64, 8),     # the ESRNet alignment shape
        (1, 32, 17, 23, 48, 2),     # odd spatial size
    ])
    def test_fused_matches_generic(self, ext, B, C, H, W, Cout, dg):
        g = torch.Generator().manual_seed(42)
        input = torch.randn(B, C, H, W, generator=g).cuda()
        offset = (torch.randn(B, dg * 18, H, W, generator=g) * 2).cuda()
        mask = torch.rand(B, dg * 9, H, W, generator=g).cuda()
        weight = (torch.randn(Cout, C, 3, 3, generator=g) * 0.2).cuda()
        bias = torch.randn(Cout, generator=g).cuda()
        fused = ext.deform_conv2d_forward_fused(input, offset, mask, weight,
                                                bias, dg)
        # generic = im2col + GEMM (itself oracle-verified)
        cols = ext.deform_im2col(input, offset, mask, 3, 3, 1, 1, 1, 1,
                                 1, 1, dg)
        ref = torch.matmul(weight.reshape(Cout, -1), cols) \
            .reshape(B, Cout, H, W) + bias.view(1, -1, 1, 1)
        err = (fused - ref).abs().max().item()
        scale = ref.abs().max().item()
        assert err / scale < 1e-5, f"rel err {err/scale}"

    def test_fused_agrees_with_public_entry(self, ext):
        # the public entry defaults to the measured-faster split path
        # (see deform_conv.hip dispatch comment); the fused kernel must
        # agree to fp32 reduction-order tolerance
        g = torch.Generator().manual_seed(1)
        input = torch.randn(2, 64, 32, 32, generator=g).cuda()
        offset = (torch.randn(2, 8 * 18, 32, 32, generator=g)).cuda()
        mask = torch.rand(2, 8 * 9, 32, 32, generator=g).cuda()
        weight = (torch.randn(64, 64, 3, 3, generator=g) * 0.2).cuda()
        bias = torch.randn(64, generator=g).cuda()
        pub = ext.deform_conv2d_forward(input, offset, mask, weight, bias,
                                        1, 1, 1, 1, 1, 1, 8)
        hook = ext.deform_conv2d_forward_fused(input, offset, mask, weight,
                                               bias, 8)
        assert torch.allclose(pub, hook, atol=1e-4, rtol=1e-4)


class TestDeformAlignGrads:
    def test_module_grads_match_cpu_oracle_batch2(self, ext):
        """DeformAlign2d backward vs the CPU autograd oracle, batch 2: the
        offset channel-slice is NON-contiguous for B>1 — this is the
        regression test for the saved-tensor contiguity bug."""
        from esr_amd.ops.dcn import DeformAlign2d
        torch.manual_seed(0)
        m = DeformAlign2d(8, 8, 3, stride=1, padding=1, deformable_groups=2)
        # non-zero offsets so the deformable path is exercised
        with torch.no_grad():
            m.conv_offset_mask.weight.normal_(0, 0.1)
            m.conv_offset_mask.bias.normal_(0, 0.5)
        x = torch.randn(2, 8, 12, 12)
        feat = torch.randn(2, 8, 12, 12)

        m_cpu = m
        out = m_cpu(x, feat)
        loss = out.square().mean()
        loss.backward()
        grads_cpu = {n: p.grad.clone() for n, p in m_cpu.named_parameters()}
        for p in m_cpu.parameters():
            p.grad = None

        m_gpu = m.cuda()
        out_gpu = m_gpu(x.cuda(), feat.cuda())
        assert torch.allclose(out_gpu.cpu(), out, atol=1e-4)
        loss_gpu = out_gpu.square().mean()
        loss_gpu.backward()
        for n, p in m_gpu.named_parameters():
            ref = grads_cpu[n]
            err = (p.grad.cpu() - ref).abs().max().item()
            scale = ref.abs().max().item() + 1e-8
            assert err / scale < 2e-3, f"{n}: rel err {err/scale:.3e}"


class TestDCNBf16:
    """bf16-native DCN path: bf16 loads, fp32 compute, vs the fp32 kernels
    on the same bf16-rounded inputs."""

    @pytest.fixture(scope="class")
    def ext(self):
        from esr_amd.ops.native import require_ext
        return require_ext()

    def test_forward_bf16_matches_fp32(self, ext):
        input, offset, mask, weight, bias = _dcn_problem(seed=11)
        b16 = [t.to(torch.bfloat16) for t in (input, offset, mask, weight)]
        out16 = ext.deform_conv2d_forward(
            *[t.contiguous() for t in b16], bias.to(torch.bfloat16),
            1, 1, 1, 1, 1, 1, 4)
        ref = ext.deform_conv2d_forward(
            *[t.to(torch.bfloat16).float().contiguous()
              for t in (input, offset, mask, weight)],
            bias.to(torch.bfloat16).float(), 1, 1, 1, 1, 1, 1, 4)
        err = (out16.float() - ref).abs().max().item()
        scale = ref.abs().max().item() + 1e-6
        assert err / scale < 2e-2, f"bf16 fwd rel err {err / scale}"

    def test_backward_bf16_matches_fp32(self, ext):
        input, offset, mask, weight, bias = _dcn_problem(seed=13)
        rounded = [t.to(torch.bfloat16).float().contiguous()
                   for t in (input, offset, mask, weight)]
        gout = torch.randn(rounded[0].shape[0], weight.shape[0],
                           rounded[0].shape[2], rounded[0].shape[3],
                           device="cuda").to(torch.bfloat16)
        g16 = ext.deform_conv2d_backward(
            *[t.to(torch.bfloat16).contiguous() for t in rounded],
            gout.contiguous(), 1, 1, 1, 1, 1, 1, 4)
        gref = ext.deform_conv2d_backward(
            *rounded, gout.float().contiguous(), 1, 1, 1, 1, 1, 1, 4)
        names = ["input", "offset", "mask", "weight", "bias"]
        for gb, gf, name in zip(g16, gref, names):
            err = (gb.float() - gf).abs().max().item()
            scale = gf.abs().max().item() + 1e-6
            assert err / scale < 3e-2, f"bf16 grad_{name} rel err {err/scale}"

    def test_module_bf16_end_to_end(self, ext):
        from esr_amd.ops.dcn import DeformAlign2d
        torch.manual_seed(7)
        m = DeformAlign2d(16, 16, 3, stride=1, padding=1,
                          deformable_groups=4).cuda().to(torch.bfloat16)
        x = torch.randn(2, 16, 24, 24, device="cuda").to(torch.bfloat16)
        f = torch.randn(2, 16, 24, 24, device="cuda").to(torch.bfloat16)
        x.requires_grad_(True)
        out = m(x, f)
        assert out.dtype == torch.bfloat16
        out.float().square().mean().backward()
        assert x.grad is not None and torch.isfinite(x.grad.float()).all()
        assert m.weight.grad is not None
