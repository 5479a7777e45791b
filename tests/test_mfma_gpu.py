"""MFMA fragment-layout probe (guide rule: asymmetric operands)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")


@requires_gpu
def test_mfma_32x32x16_bf16_layout():
  from tensor2robot_amd.ops import _t2r_hip
  torch.manual_seed(0)
  A = torch.randn(32, 16, device="cuda").to(torch.bfloat16)
  B = torch.randn(16, 32, device="cuda").to(torch.bfloat16)
  D = _t2r_hip.mfma_probe(A, B)
  ref = A.float() @ B.float()
  assert torch.allclose(D, ref, atol=2e-2, rtol=2e-2), \
      float((D - ref).abs().max())
  # Identity x asymmetric B: D == B rows exactly placed.
  I = torch.eye(32, 16, device="cuda").to(torch.bfloat16)
  B2 = torch.arange(16 * 32, device="cuda").reshape(16, 32).to(
      torch.bfloat16) / 100.0
  D2 = _t2r_hip.mfma_probe(I, B2)
  ref2 = I.float() @ B2.float()
  assert torch.allclose(D2, ref2, atol=2e-2, rtol=2e-2)


@requires_gpu
@pytest.mark.parametrize("shape", [
    # (N, C, H, W, K, R, pad)
    (2, 64, 79, 79, 64, 5, 2),    # Grasping44 block1
    (2, 64, 27, 27, 64, 3, 1),    # block2 SAME
    (2, 64, 14, 14, 64, 3, 0),    # block3 VALID
    (1, 64, 33, 17, 64, 5, 2),    # ragged edges
    (1, 32, 16, 16, 32, 3, 1),    # smaller C/K
])
def test_mfma_conv_forward_matches_torch(shape):
  import torch.nn.functional as F
  from tensor2robot_amd.ops import conv as mconv
  n, c, h, w, k, r, pad = shape
  torch.manual_seed(0)
  x = torch.randn(n, c, h, w, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last)
  weight = (torch.randn(k, c, r, r, device="cuda") * 0.1).to(
      torch.bfloat16)
  y = mconv._MFMAConvFunction.apply(x, weight, pad)
  ref = F.conv2d(x.float(), weight.float(), padding=pad)
  err = (y.float() - ref).abs().max().item()
  scale = ref.abs().max().item()
  assert err < 0.02 * max(scale, 1.0), (shape, err, scale)


@requires_gpu
def test_mfma_conv_backward_matches_torch():
  import torch.nn.functional as F
  from tensor2robot_amd.ops import conv as mconv
  torch.manual_seed(0)
  n, c, h, w, k, r, pad = 2, 64, 27, 27, 64, 3, 1
  x = torch.randn(n, c, h, w, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last).requires_grad_(True)
  weight = (torch.randn(k, c, r, r, device="cuda") * 0.1).to(
      torch.bfloat16).requires_grad_(True)
  y = mconv._MFMAConvFunction.apply(x, weight, pad)
  dy = torch.randn_like(y)
  y.backward(dy)

  x2 = x.detach().float().requires_grad_(True)
  w2 = weight.detach().float().requires_grad_(True)
  F.conv2d(x2, w2, padding=pad).backward(dy.float())
  dx_err = (x.grad.float() - x2.grad).abs().max().item()
  dx_scale = x2.grad.abs().max().item()
  assert dx_err < 0.03 * max(dx_scale, 1.0), (dx_err, dx_scale)
  dw_err = (weight.grad.float() - w2.grad).abs().max().item()
  assert dw_err < 0.03 * max(w2.grad.abs().max().item(), 1.0)


@requires_gpu
def test_mfma_conv_module_dispatch():
  from tensor2robot_amd.ops import conv as mconv
  m = mconv.MFMAConv2d(64, 64, 3, padding=1).cuda().to(
      memory_format=torch.channels_last)
  x = torch.randn(2, 64, 33, 33, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last)
  y = m(x)
  assert y.shape == (2, 64, 33, 33)
  # 5x5 falls back to torch conv (bf16 weights cast inside).
  m5 = mconv.MFMAConv2d(64, 64, 5, padding=2).cuda().to(
      memory_format=torch.channels_last)
  assert m5(x).shape == (2, 64, 33, 33)
  # Unsupported shape falls back to torch conv.
  m2 = mconv.MFMAConv2d(3, 64, 6, stride=2, padding=2).cuda()
  x2 = torch.randn(2, 3, 64, 64, device="cuda")
  assert m2(x2).shape[1] == 64


@requires_gpu
def test_pack_kernel_matches_python():
  from tensor2robot_amd.ops import conv as mconv
  from tensor2robot_amd.ops import _t2r_hip
  torch.manual_seed(0)
  w = torch.randn(64, 64, 3, 3, device="cuda").to(torch.bfloat16)
  assert torch.equal(_t2r_hip.pack_conv_w(w, False),
                     mconv.pack_weights(w))
  assert torch.equal(_t2r_hip.pack_conv_w(w, True),
                     mconv.pack_weights_bwd(w))
  w5 = torch.randn(64, 64, 5, 5, device="cuda").to(torch.bfloat16)
  assert torch.equal(_t2r_hip.pack_conv_w(w5, True),
                     mconv.pack_weights_bwd(w5))


@requires_gpu
@pytest.mark.parametrize("shape", [
    (2, 64, 79, 79, 64, 5, 2),
    (2, 64, 27, 27, 64, 3, 1),
    (1, 64, 33, 17, 64, 3, 0),
    (1, 32, 16, 16, 32, 3, 1),
])
def test_mfma_wrw_matches_torch(shape):
  import torch.nn.functional as F
  from tensor2robot_amd.ops import _t2r_hip
  n, c, h, w, k, r, pad = shape
  torch.manual_seed(0)
  x = torch.randn(n, c, h, w, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last)
  oh, ow = h + 2 * pad - r + 1, w + 2 * pad - r + 1
  dy = torch.randn(n, k, oh, ow, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last)
  dw = _t2r_hip.conv_s1_wrw(x, dy, r, r, pad)
  dw_t = dw.reshape(r, r, c, k).permute(3, 2, 0, 1).contiguous()
  # fp32 reference via autograd.
  x32 = x.float().requires_grad_(False)
  w32 = torch.zeros(k, c, r, r, device="cuda", requires_grad=True)
  F.conv2d(x32, w32, padding=pad).backward(dy.float())
  ref = w32.grad
  err = (dw_t - ref).abs().max().item()
  scale = ref.abs().max().item()
  assert err < 0.01 * max(scale, 1.0), (shape, err, scale)


@requires_gpu
def test_stem_conv_matches_torch():
  import torch.nn.functional as F
  from tensor2robot_amd.ops import _t2r_hip
  torch.manual_seed(0)
  for h, w in [(472, 472), (96, 130)]:
    x = torch.randn(2, 3, h, w, device="cuda").to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    weight = (torch.randn(64, 3, 6, 6, device="cuda") * 0.1).to(
        torch.bfloat16)
    wpk = _t2r_hip.pack_stem_w(weight)
    y = _t2r_hip.conv_stem_nhwc(x, wpk)
    ref = F.conv2d(x.float(), weight.float(), stride=2, padding=2)
    assert y.shape == ref.shape, (y.shape, ref.shape)
    err = (y.float() - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err < 0.02 * max(scale, 1.0), (h, w, err, scale)


@requires_gpu
def test_stem_conv_module_and_grads():
  from tensor2robot_amd.ops import conv as mconv
  torch.manual_seed(0)
  m = mconv.MFMAConv2d(3, 64, 6, stride=2, padding=2).cuda().to(
      torch.bfloat16).to(memory_format=torch.channels_last)
  x = torch.randn(2, 3, 96, 96, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last)
  y = m(x)
  assert y.shape == (2, 64, 48, 48)
  y.float().pow(2).mean().backward()
  assert m.weight.grad is not None
  assert torch.isfinite(m.weight.grad).all()


@pytest.mark.gpu
def test_s2d_stem_kernel_matches_reference():
  from tensor2robot_amd import ops as ops_mod
  from tensor2robot_amd.ops import conv as conv_mod
  ext = ops_mod.require_hip()
  x = torch.randn(3, 3, 472, 472, device="cuda", dtype=torch.bfloat16)
  x = x.contiguous(memory_format=torch.channels_last)
  got = ext.s2d_stem(x)
  want = conv_mod._space_to_depth_nhwc(x)
  assert got.shape == want.shape == (3, 16, 236, 236)
  assert got.is_contiguous(memory_format=torch.channels_last)
  assert torch.equal(got, want)


@requires_gpu
@pytest.mark.parametrize("shape", [
    (2, 64, 79, 79, 64, 5, 2),
    (2, 64, 27, 27, 64, 3, 1),
    (1, 64, 33, 17, 64, 3, 0),
    (3, 64, 78, 78, 64, 5, 2),
])
def test_mfma_wrw2_matches_torch(shape):
  import torch.nn.functional as F
  from tensor2robot_amd.ops import _t2r_hip
  n, c, h, w, k, r, pad = shape
  torch.manual_seed(0)
  x = torch.randn(n, c, h, w, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last)
  oh, ow = h + 2 * pad - r + 1, w + 2 * pad - r + 1
  dy = torch.randn(n, k, oh, ow, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last)
  dw = _t2r_hip.conv_s1_wrw2(x, dy, r, r, pad)
  dw_t = dw.reshape(r, r, c, k).permute(3, 2, 0, 1).contiguous()
  x32 = x.float().requires_grad_(False)
  w32 = torch.zeros(k, c, r, r, device="cuda", requires_grad=True)
  F.conv2d(x32, w32, padding=pad).backward(dy.float())
  ref = w32.grad
  err = (dw_t - ref).abs().max().item()
  scale = ref.abs().max().item()
  assert err < 0.01 * max(scale, 1.0), (shape, err, scale)


@requires_gpu
def test_pack_conv_w_pair_matches_single():
  from tensor2robot_amd.ops import _t2r_hip
  for shape in [(64, 64, 3, 3), (64, 64, 5, 5), (32, 64, 3, 3)]:
    w = torch.randn(*shape, device="cuda").to(torch.bfloat16)
    outf, outb = _t2r_hip.pack_conv_w_pair(w)
    assert torch.equal(outf, _t2r_hip.pack_conv_w(w, False))
    assert torch.equal(outb, _t2r_hip.pack_conv_w(w, True))


@requires_gpu
def test_fused_spatial_softmax_matches_torch():
  import os
  from tensor2robot_amd.layers import spatial_softmax as ss
  torch.manual_seed(3)
  x = torch.randn(4, 64, 13, 17, device="cuda") * 3.0
  xb = x.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
  # fp32 reference math on the SAME bf16-quantized input (isolates
  # kernel numerics from input quantization).
  x = xb.float()
  mod = ss.SpatialSoftmax(temperature=0.7)

  # fp32 torch reference (fused path disabled).
  os.environ["T2R_DISABLE_FUSED_SPATIAL_SOFTMAX"] = "1"
  try:
    xr = x.clone().requires_grad_(True)
    pts_ref, map_ref = mod(xr)
    (pts_ref.sum() + (map_ref * 0.01).sum()).backward()
  finally:
    del os.environ["T2R_DISABLE_FUSED_SPATIAL_SOFTMAX"]

  xf = xb.clone().requires_grad_(True)
  pts, smap = mod(xf)
  assert pts.shape == (4, 128) and smap.shape == (4, 64, 13, 17)
  assert (pts.float() - pts_ref).abs().max().item() < 2e-2
  assert (smap.float() - map_ref).abs().max().item() < 2e-2
  (pts.sum() + (smap.float() * 0.01).sum()).backward()
  scale = xr.grad.abs().max().item()
  assert (xf.grad.float() - xr.grad).abs().max().item() < 0.05 * scale

  # points-only gradient path (dmap is None).
  xg = xb.clone().requires_grad_(True)
  pts2, _ = mod(xg)
  pts2.sum().backward()
  xr2 = x.clone().requires_grad_(True)
  os.environ["T2R_DISABLE_FUSED_SPATIAL_SOFTMAX"] = "1"
  try:
    p_ref2, _ = mod(xr2)
    p_ref2.sum().backward()
  finally:
    del os.environ["T2R_DISABLE_FUSED_SPATIAL_SOFTMAX"]
  scale2 = xr2.grad.abs().max().item()
  assert (xg.grad.float() - xr2.grad).abs().max().item() < 0.05 * scale2


@requires_gpu
@pytest.mark.parametrize("shape", [
    (2, 64, 79, 79, 64, 5, 2),
    (2, 64, 27, 27, 64, 3, 1),
    (3, 64, 78, 78, 64, 5, 2),
])
def test_mfma_wrw3_matches_torch(shape):
  import torch.nn.functional as F
  from tensor2robot_amd.ops import _t2r_hip
  n, c, h, w, k, r, pad = shape
  torch.manual_seed(0)
  x = torch.randn(n, c, h, w, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last)
  oh, ow = h + 2 * pad - r + 1, w + 2 * pad - r + 1
  dy = torch.randn(n, k, oh, ow, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last)
  dw = _t2r_hip.conv_s1_wrw3(x, dy, r, r, pad)
  dw_t = dw.reshape(r, r, c, k).permute(3, 2, 0, 1).contiguous()
  x32 = x.float()
  w32 = torch.zeros(k, c, r, r, device="cuda", requires_grad=True)
  F.conv2d(x32, w32, padding=pad).backward(dy.float())
  ref = w32.grad
  err = (dw_t - ref).abs().max().item()
  scale = ref.abs().max().item()
  assert err < 0.01 * max(scale, 1.0), (shape, err, scale)


@requires_gpu
def test_spatial_softmax_large_map_fallback():
  """Maps past the 4096-pixel fused cap use the torch composition on
  GPU with matching numerics."""
  from tensor2robot_amd.layers import spatial_softmax as ss
  torch.manual_seed(1)
  mod = ss.SpatialSoftmax()
  x = torch.randn(2, 8, 80, 80, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last)
  assert not mod._fused_supported(x)
  points, smap = mod(x)
  assert points.shape == (2, 16) and smap.shape == (2, 8, 80, 80)
  ref_pts, _ = mod(x.float())
  assert (points.float() - ref_pts).abs().max().item() < 2e-2


@requires_gpu
@pytest.mark.parametrize("shape", [
    (2, 64, 37, 37, 64, 3, 1),
    (2, 64, 78, 78, 64, 5, 2),
    (1, 64, 33, 29, 64, 3, 1),     # non-square spatial + odd edges
    (3, 64, 16, 16, 64, 5, 2),     # window == tile edge case
])
def test_mfma_wrw4_matches_torch(shape):
  """v4 numerics: the G17 tr_b16 trap (base = 2/4/6 mod 8 shorts reads
  the 8-aligned address's data with NO stall) makes this the first
  gate on any layout change — run before timing anything."""
  import torch.nn.functional as F
  from tensor2robot_amd.ops import _t2r_hip
  n, c, h, w, k, r, pad = shape
  torch.manual_seed(0)
  x = torch.randn(n, c, h, w, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last)
  oh, ow = h + 2 * pad - r + 1, w + 2 * pad - r + 1
  dy = torch.randn(n, k, oh, ow, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last)
  dw = _t2r_hip.conv_s1_wrw4(x, dy, r, r, pad)
  dw_t = dw.float()  # v4 emits [K,C,R,S] bf16 directly
  x32 = x.float()
  w32 = torch.zeros(k, c, r, r, device="cuda", requires_grad=True)
  F.conv2d(x32, w32, padding=pad).backward(dy.float())
  ref = w32.grad
  err = (dw_t - ref).abs().max().item()
  scale = ref.abs().max().item()
  assert err < 0.01 * max(scale, 1.0), (shape, err, scale)


@requires_gpu
@pytest.mark.parametrize("shape", [
    # (n, c, h, w, k, r, stride, pad) — ResNet-family GEMM-conv shapes
    (4, 64, 25, 25, 64, 3, 1, 1),
    (4, 64, 25, 25, 128, 3, 2, 1),     # downsample
    (4, 128, 13, 13, 128, 3, 1, 1),
    (4, 256, 7, 7, 512, 1, 1, 0),      # 1x1 projection
    (4, 64, 25, 25, 256, 1, 1, 0),     # bottleneck expand
    (2, 512, 4, 4, 512, 3, 1, 1),      # deep tiny-spatial
    (2, 64, 14, 14, 128, 1, 2, 0),     # 1x1 stride-2 projection
])
def test_gemm_conv_matches_torch(shape):
  """GEMM-conv (im2col + rocBLAS) numerics vs fp32 reference."""
  import torch.nn.functional as F
  from tensor2robot_amd.ops import gemm_conv
  n, c, h, w, k, r, stride, pad = shape
  torch.manual_seed(0)
  x = torch.randn(n, c, h, w, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last).requires_grad_(True)
  wt = (torch.randn(k, c, r, r, device="cuda") * 0.05).to(torch.bfloat16)
  wt.requires_grad_(True)
  y = gemm_conv.gemm_conv2d(x, wt, (stride, stride), (pad, pad))
  oh = (h + 2 * pad - r) // stride + 1
  assert y.shape == (n, k, oh, oh)
  dy = torch.randn_like(y.float()).to(torch.bfloat16)
  y.backward(dy)

  x32 = x.detach().float().requires_grad_(True)
  w32 = wt.detach().float().requires_grad_(True)
  y32 = F.conv2d(x32, w32, stride=stride, padding=pad)
  y32.backward(dy.float())

  def relerr(a, b):
    return (a.float() - b).abs().max().item() / max(
        b.abs().max().item(), 1e-6)

  assert relerr(y, y32) < 0.02, ("y", shape, relerr(y, y32))
  assert relerr(x.grad, x32.grad) < 0.03, ("dx", shape)
  assert relerr(wt.grad, w32.grad) < 0.03, ("dw", shape)


@requires_gpu
def test_gemm_conv_cpad_stem_matches_torch():
  """RGB 7x7/2 stem via the channel-padded GEMM path."""
  import torch.nn.functional as F
  from tensor2robot_amd.ops import gemm_conv
  torch.manual_seed(0)
  x = torch.randn(4, 3, 100, 100, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last)
  wt = (torch.randn(64, 3, 7, 7, device="cuda") * 0.05).to(
      torch.bfloat16).requires_grad_(True)
  y = gemm_conv.gemm_conv2d_cpad(x, wt, (2, 2), (3, 3))
  dy = torch.randn_like(y.float()).to(torch.bfloat16)
  y.backward(dy)
  w32 = wt.detach().float().requires_grad_(True)
  y32 = F.conv2d(x.float(), w32, stride=2, padding=3)
  y32.backward(dy.float())
  assert (y.float() - y32).abs().max().item() < \
      0.02 * y32.abs().max().item()
  assert (wt.grad.float() - w32.grad).abs().max().item() < \
      0.03 * max(w32.grad.abs().max().item(), 1.0)


@requires_gpu
@pytest.mark.parametrize("shape", [
    # n, c, h, w, k, pad   (3x3 stride-1 only)
    (2, 128, 59, 59, 128, 1),    # G2V/ResNet50 layer2
    (2, 256, 30, 30, 256, 1),    # layer3
    (2, 512, 15, 15, 512, 1),    # layer4
    (2, 128, 25, 25, 128, 1),    # BC-Z ResNet18 @100^2
    (2, 96, 20, 20, 64, 1),      # C%32 only, K=64
    (2, 128, 17, 19, 128, 0),    # VALID pad, non-square map
])
def test_bigc_conv_matches_torch(shape):
  """C-chunked MFMA 3x3 (conv_s1_big.hip) vs fp32 reference:
  forward, dgrad (chunk kernel on flipped pack) and dw (GEMM)."""
  import torch.nn.functional as F
  from tensor2robot_amd.ops import conv as conv_mod
  n, c, h, w, k, pad = shape
  torch.manual_seed(0)
  x = torch.randn(n, c, h, w, device="cuda").to(torch.bfloat16) \
      .contiguous(memory_format=torch.channels_last)
  need_dx = c % 64 == 0 and k % 64 == 0
  x.requires_grad_(need_dx)
  wt = (torch.randn(k, c, 3, 3, device="cuda") * 0.05).to(
      torch.bfloat16).requires_grad_(True)
  y = conv_mod._BigCConvFunction.apply(x, wt, pad)
  oh, ow = h + 2 * pad - 2, w + 2 * pad - 2
  assert y.shape == (n, k, oh, ow)
  dy = torch.randn_like(y.float()).to(torch.bfloat16)
  y.backward(dy)

  x32 = x.detach().float().requires_grad_(True)
  w32 = wt.detach().float().requires_grad_(True)
  y32 = F.conv2d(x32, w32, stride=1, padding=pad)
  y32.backward(dy.float())

  def relerr(a, b):
    return (a.float() - b).abs().max().item() / max(
        b.abs().max().item(), 1e-6)

  assert relerr(y, y32) < 0.02, ("y", shape, relerr(y, y32))
  if need_dx:
    assert relerr(x.grad, x32.grad) < 0.03, ("dx", shape,
                                             relerr(x.grad, x32.grad))
  assert relerr(wt.grad, w32.grad) < 0.03, ("dw", shape,
                                            relerr(wt.grad, w32.grad))
