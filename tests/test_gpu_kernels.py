"""GPU numerics tests: hand-written gfx950 kernels vs plain PyTorch fp32
references (asymmetric inputs so transposed-output bugs can't pass)."""

import os

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs MI355X"
)


def _rel_err(out, ref):
    return ((out.float() - ref).abs().max() / (ref.abs().max() + 1e-6)).item()


@needs_gpu
class TestGemm:
    def test_gemm_nt_matches_fp32(self):
        from beta9_amd.ops import gemm as ops

        assert ops.has_native(), "native kernels must load on GPU"
        torch.manual_seed(0)
        M, N, K = 256, 384, 512
        a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.5
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.5
        out = ops.gemm_nt(a, w)
        ref = a.float() @ w.float().t()
        assert out.shape == (M, N)
        # bf16 in / fp32 accumulate: tolerance scales with sqrt(K)*eps_bf16
        assert _rel_err(out, ref) < 0.02, f"rel err {_rel_err(out, ref)}"

    def test_gemm_asymmetric_catches_transpose(self):
        from beta9_amd.ops import gemm as ops

        M, N, K = 128, 256, 128
        a = torch.zeros(M, K, dtype=torch.bfloat16, device="cuda")
        w = torch.zeros(N, K, dtype=torch.bfloat16, device="cuda")
        a[3, 5] = 2.0
        w[7, 5] = 3.0
        out = ops.gemm_nt(a, w)
        assert abs(out[3, 7].item() - 6.0) < 1e-3
        assert out.float().abs().sum().item() == pytest.approx(6.0, abs=1e-2)

    def test_gemm_padded_m(self):
        from beta9_amd.ops import gemm as ops

        a = torch.randn(100, 256, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(128, 256, dtype=torch.bfloat16, device="cuda")
        out = ops.gemm_nt(a, w)
        ref = a.float() @ w.float().t()
        assert out.shape == (100, 128)
        assert _rel_err(out, ref) < 0.02

    def test_gemm_large_shapes(self):
        from beta9_amd.ops import gemm as ops

        for (M, N, K) in [(512, 4096, 4096), (4096, 4096, 4096)]:
            a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.1
            w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.1
            out = ops.gemm_nt(a, w)
            ref = a.float() @ w.float().t()
            assert _rel_err(out, ref) < 0.02

    def test_fused_gate_up_silu(self):
        from beta9_amd.ops import gemm as ops

        torch.manual_seed(1)
        M, N, K = 256, 384, 512
        a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.3
        wg = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.3
        wu = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.3
        out = ops.mlp_gate_up(a, wg, wu)
        g = a.float() @ wg.float().t()
        u = a.float() @ wu.float().t()
        ref = torch.nn.functional.silu(g) * u
        assert out.shape == (M, N)
        assert _rel_err(out, ref) < 0.03

    def test_fused_gate_up_padded_m(self):
        from beta9_amd.ops import gemm as ops

        a = torch.randn(100, 256, dtype=torch.bfloat16, device="cuda") * 0.3
        wg = torch.randn(128, 256, dtype=torch.bfloat16, device="cuda") * 0.3
        wu = torch.randn(128, 256, dtype=torch.bfloat16, device="cuda") * 0.3
        out = ops.mlp_gate_up(a, wg, wu)
        ref = torch.nn.functional.silu(a.float() @ wg.float().t()) * (a.float() @ wu.float().t())
        assert out.shape == (100, 128)
        assert _rel_err(out, ref) < 0.03

    def test_silu_mul(self):
        from beta9_amd.ops import gemm as ops

        g = torch.randn(1000, dtype=torch.bfloat16, device="cuda")
        u = torch.randn(1000, dtype=torch.bfloat16, device="cuda")
        out = ops.silu_mul(g, u)
        ref = torch.nn.functional.silu(g.float()) * u.float()
        assert _rel_err(out, ref) < 0.02

    def test_rmsnorm(self):
        from beta9_amd.ops import gemm as ops

        x = torch.randn(32, 1024, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(1024, dtype=torch.bfloat16, device="cuda")
        out = ops.rmsnorm(x, w)
        xf = x.float()
        ref = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-6) * w.float()
        assert _rel_err(out, ref) < 0.03


@needs_gpu
class TestArena:
    def test_upload_download_roundtrip(self):
        from beta9_amd.ops import hipsnap

        a = hipsnap.Arena(device=0, size_bytes=64 << 20, staging_bytes=8 << 20)
        assert a.native
        data = os.urandom(20 << 20)  # > staging half: exercises the pipeline
        off = a.alloc("blob", len(data))
        a.upload(off, data)
        back = a.download(off, len(data))
        assert back == data

    def test_file_streaming(self, tmp_path):
        from beta9_amd.ops import hipsnap

        a = hipsnap.Arena(device=0, size_bytes=64 << 20, staging_bytes=8 << 20)
        data = os.urandom(12 << 20)
        src = tmp_path / "weights.bin"
        src.write_bytes(data)
        off = a.alloc("w", len(data))
        a._impl.upload_file(off, str(src), 0, len(data))
        dst = tmp_path / "snap.bin"
        a._impl.download_file(off, str(dst), len(data), False)
        assert dst.read_bytes() == data

    def test_snapshot_restore(self, tmp_path):
        import torch

        from beta9_amd.ops import hipsnap

        a = hipsnap.Arena(device=0, size_bytes=32 << 20, staging_bytes=4 << 20)
        t1 = a.tensor("w1", (1024, 512), dtype="bfloat16")
        t2 = a.tensor("w2", (2048,), dtype="float32")
        t1.normal_()
        t2.uniform_()
        c1, c2 = t1.float().sum().item(), t2.sum().item()
        info = hipsnap.snapshot(a, str(tmp_path), "ckpt1")
        assert info["bytes"] == t1.numel() * 2 + t2.numel() * 4

        b = hipsnap.Arena(device=0, size_bytes=32 << 20, staging_bytes=4 << 20)
        assert hipsnap.restore_snapshot(b, "ckpt1", store_dir=str(tmp_path))
        r1 = b.tensor_like_region("w1") if hasattr(b, "tensor_like_region") else None
        # regions were re-registered: view them back
        regions = {n: (o, s) for (n, o, s) in b.regions()}
        o1, s1 = regions["w1"]
        v1 = b._impl.tensor_view(o1, s1).view(torch.bfloat16)
        assert abs(v1.float().sum().item() - c1) < 1.0
        o2, s2 = regions["w2"]
        v2 = b._impl.tensor_view(o2, s2).view(torch.float32)
        assert abs(v2.sum().item() - c2) < 1e-2


@needs_gpu
def test_serving_model_forward():
    from beta9_amd.models.mlp import SyntheticServingModel

    m = SyntheticServingModel(hidden=1024, inter=2048, layers=2)
    r = m.infer(batch_tokens=128)
    assert r["checksum"] == r["checksum"]
    assert r["tflops"] > 0


@needs_gpu
def test_transformer_block_forward():
    from beta9_amd.models.transformer import TransformerServingModel

    m = TransformerServingModel(hidden=1024, inter=2048, layers=2, n_heads=8, n_kv_heads=2)
    r = m.infer(batch_tokens=128)
    assert r["checksum"] == r["checksum"] and r["tflops"] > 0
    # numerics sanity: forward produces finite values at realistic magnitudes
    assert 0 < r["checksum"] < 1e3


@pytest.mark.gpu
def test_gemm_residual_numerics():
    """Fused residual epilogue vs fp32 reference, plain and split-K paths."""
    import torch

    from beta9_amd.ops import gemm as g

    ext = g._load()
    assert ext is not None
    for (M, N, K) in [(256, 4096, 14336), (128, 128, 64), (512, 512, 512)]:
        a = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
        w = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
        r = (torch.randn(M, N, device="cuda") * 0.5).to(torch.bfloat16)
        ref = r.float() + a.float() @ w.float().t()
        for sk in (-1, 1):
            out = ext.gemm_bf16_nt_residual(a, w, r, False, sk)
            err = ((out.float() - ref).abs().max() / ref.abs().max()).item()
            assert err < 0.02, (M, N, K, sk, err)
    # padded-M wrapper path
    a = (torch.randn(100, 256, device="cuda") * 0.5).to(torch.bfloat16)
    w = (torch.randn(128, 256, device="cuda") * 0.5).to(torch.bfloat16)
    r = (torch.randn(100, 128, device="cuda") * 0.5).to(torch.bfloat16)
    out = g.gemm_nt_residual(a, w, r)
    ref = r.float() + a.float() @ w.float().t()
    assert ((out.float() - ref).abs().max() / ref.abs().max()).item() < 0.02


@pytest.mark.gpu
def test_gate_up_fp8_numerics():
    """FP8 w8a8 fused gate-up vs the dequantized fp32 reference."""
    import torch

    from beta9_amd.ops import gemm as g

    torch.manual_seed(3)
    a = (torch.randn(256, 512, device="cuda") * 0.5).to(torch.bfloat16)
    wg = (torch.randn(512, 512, device="cuda") * 0.5).to(torch.bfloat16)
    wu = (torch.randn(512, 512, device="cuda") * 0.5).to(torch.bfloat16)
    aq, a_s = g.quantize_fp8_tensor(a)
    gq, g_s = g.quantize_fp8_rowwise(wg)
    uq, u_s = g.quantize_fp8_rowwise(wu)
    out = g.mlp_gate_up_fp8(aq, gq, uq, g_s, u_s, a_s)
    adq = aq.view(torch.float8_e4m3fn).float() * a_s
    gdq = gq.view(torch.float8_e4m3fn).float() * g_s[:, None]
    udq = uq.view(torch.float8_e4m3fn).float() * u_s[:, None]
    ref = torch.nn.functional.silu(adq @ gdq.t()) * (adq @ udq.t())
    err = ((out.float() - ref).abs().max() / ref.abs().max()).item()
    assert err < 0.02, err


@pytest.mark.gpu
def test_serving_model_fp8_backend():
    """B9_GEMM_BACKEND=fp8 runs the w8a8 serving path end to end and stays
    close to the bf16 output (quantization-level error)."""
    import os

    import torch

    from beta9_amd.models.mlp import SyntheticServingModel

    m = SyntheticServingModel(512, 1024, 2, "cuda:0", use_arena=False)
    x = torch.randn(64, 512, dtype=torch.bfloat16, device="cuda:0")
    old = os.environ.get("B9_GEMM_BACKEND")
    try:
        os.environ["B9_GEMM_BACKEND"] = "native"
        ref = m.forward(x).float()
        os.environ["B9_GEMM_BACKEND"] = "fp8"
        out = m.forward(x).float()
    finally:
        if old is None:
            os.environ.pop("B9_GEMM_BACKEND", None)
        else:
            os.environ["B9_GEMM_BACKEND"] = old
    rel = ((out - ref).abs().mean() / ref.abs().mean()).item()
    assert rel < 0.1, rel  # quantization-level deviation, not garbage


@pytest.mark.gpu
def test_rmsnorm_fp8_fused_quant():
    """rmsnorm with fused e4m3 output vs separate rmsnorm + quantize."""
    import torch

    from beta9_amd.ops import gemm as g

    torch.manual_seed(5)
    x = (torch.randn(64, 512, device="cuda") * 0.5).to(torch.bfloat16)
    w = torch.ones(512, dtype=torch.bfloat16, device="cuda")
    h = g.rmsnorm(x, w)
    scale = (h.float().abs().amax() / g.E4M3_MAX).item()
    ref = (h.float() / scale).to(torch.float8_e4m3fn).float()
    out = g.rmsnorm_fp8_out(x, w, scale).view(torch.float8_e4m3fn).float()
    # dequantized output close to the bf16 rmsnorm (e4m3 steps scale with
    # magnitude, so only a relative check is meaningful)
    rel = ((out * scale - h.float()).abs().mean() / h.float().abs().mean()).item()
    assert rel < 0.05, rel
    # and close to the reference quantization (within ~one quant step rel.)
    rel2 = ((out - ref).abs().mean() / (ref.abs().mean() + 1e-6)).item()
    assert rel2 < 0.08, rel2


@pytest.mark.gpu
def test_down_fp8_numerics():
    """FP8 w8a8 split-K down-projection + fused residual/dequant vs the
    dequantized fp32 reference (VERDICT round-1 item #8)."""
    import torch

    from beta9_amd.ops import gemm as g

    torch.manual_seed(5)
    M, K, N = 256, 1024, 512  # K%128==0, N%64==0
    a = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
    w = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
    r = torch.randn(M, N, device="cuda").to(torch.bfloat16)
    aq, a_s = g.quantize_fp8_tensor(a)
    wq, w_s = g.quantize_fp8_rowwise(w)
    out = g.mlp_down_fp8(aq, wq, w_s, a_s, r)
    adq = aq.view(torch.float8_e4m3fn).float() * a_s
    wdq = wq.view(torch.float8_e4m3fn).float() * w_s[:, None]
    ref = r.float() + adq @ wdq.t()
    err = ((out.float() - ref).abs().max() / ref.abs().max()).item()
    assert err < 0.02, err
    # no-residual path
    out2 = g.mlp_down_fp8(aq, wq, w_s, a_s, None)
    ref2 = adq @ wdq.t()
    err2 = ((out2.float() - ref2).abs().max() / ref2.abs().max()).item()
    assert err2 < 0.02, err2


@pytest.mark.gpu
def test_gate_up_fp8_q8_epilogue():
    """Q8 epilogue emits e4m3 of silu(g)*u/act2 matching the bf16 kernel's
    output after quantization."""
    import torch

    from beta9_amd.ops import gemm as g

    torch.manual_seed(7)
    a = (torch.randn(128, 512, device="cuda") * 0.5).to(torch.bfloat16)
    wg = (torch.randn(512, 512, device="cuda") * 0.5).to(torch.bfloat16)
    wu = (torch.randn(512, 512, device="cuda") * 0.5).to(torch.bfloat16)
    aq, a_s = g.quantize_fp8_tensor(a)
    gq, g_s = g.quantize_fp8_rowwise(wg)
    uq, u_s = g.quantize_fp8_rowwise(wu)
    bf = g.mlp_gate_up_fp8(aq, gq, uq, g_s, u_s, a_s)
    act2 = (bf.float().abs().amax() * 1.1 / g.E4M3_MAX).item()
    q8 = g.mlp_gate_up_fp8_q8(aq, gq, uq, g_s, u_s, a_s, act2)
    deq = q8.view(torch.float8_e4m3fn).float() * act2
    err = ((deq - bf.float()).abs().max() / bf.float().abs().max()).item()
    assert err < 0.05, err  # one extra e4m3 rounding


@pytest.mark.gpu
def test_fp8_backend_steady_state_full_chain():
    """Second fp8 forward call runs the FULL w8a8 chain (gate-up Q8 ->
    fp8 down + fused residual) and stays at quantization-level error."""
    import os

    import torch

    from beta9_amd.models.mlp import SyntheticServingModel

    m = SyntheticServingModel(512, 1024, 2, "cuda:0", use_arena=False)
    x = torch.randn(64, 512, dtype=torch.bfloat16, device="cuda:0")
    old = os.environ.get("B9_GEMM_BACKEND")
    try:
        os.environ["B9_GEMM_BACKEND"] = "native"
        ref = m.forward(x).float()
        os.environ["B9_GEMM_BACKEND"] = "fp8"
        m.forward(x)  # calibration call (bf16 down)
        for blk in m.blocks:
            assert blk._act2_scale is not None
        out = m.forward(x).float()  # steady state: full fp8 chain
    finally:
        if old is None:
            os.environ.pop("B9_GEMM_BACKEND", None)
        else:
            os.environ["B9_GEMM_BACKEND"] = old
    rel = ((out - ref).abs().mean() / ref.abs().mean()).item()
    assert rel < 0.12, rel
